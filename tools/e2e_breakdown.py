#!/usr/bin/env python3
"""Per-phase timing of the e2e ingest tick (bench window 1) to locate host
stalls: fresh-encode, payload stamp, H2D copies, device tick, result D2H.
Run on a GPU box; writes JSON lines to stdout."""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from cordum_amd.ops.pipeline import DevicePipeline


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 40
    pipe = DevicePipeline(device="cuda:0", batch_size=16384, n_local_workers=1000,
                          n_rules=1024, n_batches=2, use_mfma=False)
    pipe.ensure_e2e()
    # warmup (graph capture out of the measured loop) + bench's GC discipline
    for _ in range(3):
        pipe.tick_e2e()
    import gc

    gc.collect()
    gc.freeze()
    gc.disable()
    rows = []
    for s in range(steps):
        t = {}
        t0 = time.perf_counter()
        pipe._e2e_enc.fresh(pipe._e2e_host)
        t["fresh"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        payload = pipe._e2e_payloads[s % len(pipe._e2e_payloads)]
        payload.view(pipe.B, pipe.payload_words)[:, 0] = s
        t["stamp"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        jb = pipe.batches[0]
        h = pipe._e2e_host
        jb.any_bits.copy_(h.any_bits, non_blocking=True)
        jb.all_bits.copy_(h.all_bits, non_blocking=True)
        jb.secrets.copy_(h.secrets, non_blocking=True)
        jb.mcp_bits.copy_(h.mcp_bits, non_blocking=True)
        jb.mcp_used.copy_(h.mcp_used, non_blocking=True)
        pipe.payloads[0].copy_(payload, non_blocking=True)
        t["h2d_submit"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        pipe._tick = 0
        st = pipe.tick()
        t["tick"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        _ = pipe.res_sums.cpu()
        _ = pipe.out_decision.cpu()
        torch.cuda.synchronize()
        t["d2h"] = time.perf_counter() - t0
        t["completed"] = st.completed
        rows.append(t)
        print(json.dumps({k: (round(v * 1e3, 3) if isinstance(v, float) else v)
                          for k, v in t.items()}))

    import statistics as stats
    for k in ("fresh", "stamp", "h2d_submit", "tick", "d2h"):
        vals = [r[k] * 1e3 for r in rows]
        print(f"# {k}: p50={stats.median(vals):.3f}ms mean={sum(vals)/len(vals):.3f}ms "
              f"max={max(vals):.3f}ms", file=sys.stderr)


if __name__ == "__main__":
    main()
