#!/usr/bin/env python3
"""Within-probe A/B of the two K1 implementations (bitset-VALU vs MFMA) at
several (jobs, rules) shapes. Interleaved timing rounds in one process
(guide §5.4 rule 24); prints one JSON line per shape."""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import time


def main():
    import torch

    if not torch.cuda.is_available():
        print("needs GPU", file=sys.stderr)
        return 1
    from cordum_amd.ops import get_ext
    from cordum_amd.ops.pipeline import encode_synthetic_jobs, make_synthetic_policy
    from cordum_amd.ops.policy_compile import compile_policy
    from cordum_amd.ops.policy_mfma import pack_jobs_mfma, pack_policy_mfma

    ext = get_ext(required=True)
    d = torch.device("cuda:0")

    for (J, R) in [(4096, 1024), (4096, 16384), (8192, 65536), (8192, 100000)]:
        policy = make_synthetic_policy(R, vocab=48, deny_frac=0.02, seed=11)
        c = compile_policy(policy, words=1)
        assert c.exact
        jobs = encode_synthetic_jobs(c, J, vocab=48, seed=12)
        cd = c.to(d)
        jb = jobs.to(d)
        mp = pack_policy_mfma(c).to(d)
        a_pack, jsec = pack_jobs_mfma(jobs)
        a_pack, jsec = a_pack.to(d), jsec.to(d)

        def run_bitset():
            return ext.policy_first_match(
                cd.any_masks, cd.all_masks, cd.secrets, cd.mcp_masks, cd.mcp_any,
                jb.any_bits, jb.all_bits, jb.secrets, jb.mcp_bits, jb.mcp_used, 0)

        def run_mfma():
            return ext.policy_first_match_mfma(a_pack, mp.b_pack, mp.cards,
                                               mp.secrets, jsec, mp.tile_dims, J, R)

        o1, o2 = run_bitset(), run_mfma()
        torch.cuda.synchronize()
        assert torch.equal(o1, o2), f"variant mismatch at J={J} R={R}"

        times = {"bitset": [], "mfma": []}
        for _ in range(10):  # interleaved rounds
            for name, fn in (("bitset", run_bitset), ("mfma", run_mfma)):
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                fn()
                torch.cuda.synchronize()
                times[name].append(time.perf_counter() - t0)
        med = {k: sorted(v)[len(v) // 2] * 1e3 for k, v in times.items()}
        rule_bytes = {"bitset": 17 * 8, "mfma": 9 * 64}
        print(json.dumps({
            "jobs": J, "rules": R,
            "bitset_ms": round(med["bitset"], 3),
            "mfma_ms": round(med["mfma"], 3),
            "mfma_over_bitset": round(med["mfma"] / med["bitset"], 2),
            "bytes_per_rule": rule_bytes,
            "evals_per_sec_bitset": round(J * R / (med["bitset"] / 1e3)),
        }))
    return 0


if __name__ == "__main__":
    sys.exit(main())
