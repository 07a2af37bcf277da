"""Policy compiler tests: the compiled bitset evaluator (torch reference for
the K1 HIP kernel) must agree with the host first-match evaluator
(safety/policy.py, oracle safety_policy.go) on randomized policies+inputs."""
import random

import pytest

from cordum_amd.ops.policy_compile import (
    CompiledPolicy,
    JobEncoder,
    compile_policy,
    first_match_reference,
)
from cordum_amd.safety import policy as pol


def random_policy(rng, n_rules=40, vocab=20):
    tenants = [f"t{i}" for i in range(vocab)]
    topics = [f"job.a{i}" for i in range(vocab)] + ["job.x.*", "job.*", "job.y?z"]
    caps = [f"cap{i}" for i in range(vocab)]
    tags = [f"tag{i}" for i in range(vocab)]
    reqs = [f"req{i}" for i in range(vocab)]
    rules = []
    for i in range(n_rules):
        m = pol.PolicyMatch()
        if rng.random() < 0.5:
            m.tenants = rng.sample(tenants, rng.randint(1, 3))
        if rng.random() < 0.5:
            m.topics = rng.sample(topics, rng.randint(1, 3))
        if rng.random() < 0.3:
            m.capabilities = rng.sample(caps, rng.randint(1, 2))
        if rng.random() < 0.3:
            m.risk_tags = rng.sample(tags, rng.randint(1, 3))
        if rng.random() < 0.3:
            m.requires = rng.sample(reqs, rng.randint(1, 2))
        if rng.random() < 0.2:
            m.labels = {f"k{rng.randint(0, 5)}": f"v{rng.randint(0, 3)}"}
        if rng.random() < 0.2:
            m.secrets_present = rng.random() < 0.5
        if rng.random() < 0.15:
            m.mcp = pol.MCPPolicy(
                deny_tools=rng.sample(["shell", "exec", "rm"], rng.randint(1, 2)),
                allow_servers=rng.sample(["srv1", "srv2"], rng.randint(0, 2)),
            )
        decision = rng.choice(["allow", "deny", "require_approval", "allow_with_constraints", "throttle"])
        rules.append(pol.PolicyRule(id=f"r{i}", match=m, decision=decision))
    return pol.SafetyPolicy(version="v1", rules=rules)


def random_input(rng, vocab=20):
    inp = pol.PolicyInput(
        tenant=f"t{rng.randint(0, vocab - 1)}",
        topic=rng.choice([f"job.a{rng.randint(0, vocab - 1)}", "job.x.deep", "job.yqz", "job.other"]),
        capability=rng.choice(["", f"cap{rng.randint(0, vocab - 1)}"]),
        risk_tags=rng.sample([f"tag{i}" for i in range(vocab)], rng.randint(0, 3)),
        requires=rng.sample([f"req{i}" for i in range(vocab)], rng.randint(0, 3)),
        labels={f"k{rng.randint(0, 5)}": f"v{rng.randint(0, 3)}"} if rng.random() < 0.5 else {},
        secrets_present=rng.random() < 0.3,
    )
    if rng.random() < 0.3:
        inp.mcp = pol.MCPRequest(
            server=rng.choice(["", "srv1", "srv2", "srv3"]),
            tool=rng.choice(["", "shell", "search"]),
        )
    return inp


@pytest.mark.parametrize("seed", [1, 2, 3, 4])
def test_compiled_matches_host_evaluator(seed):
    rng = random.Random(seed)
    policy = random_policy(rng)
    compiled = compile_policy(policy, words=1)
    assert compiled.exact, "test vocab should fit one word"
    encoder = JobEncoder(compiled)
    inputs = [random_input(rng) for _ in range(300)]
    batch = encoder.encode(inputs)
    got = first_match_reference(compiled, batch)
    for j, inp in enumerate(inputs):
        want = policy.evaluate(inp)
        rule_idx = int(got[j])
        if rule_idx < 0:
            assert want.rule_id == "", f"job {j}: host matched {want.rule_id}, compiled matched none ({inp})"
        else:
            assert compiled.rules[rule_idx].id == want.rule_id, (
                f"job {j}: host={want.rule_id} compiled={compiled.rules[rule_idx].id} ({inp})"
            )


def test_vocab_overflow_marks_inexact():
    rules = [
        pol.PolicyRule(id=f"r{i}", match=pol.PolicyMatch(tenants=[f"tenant-{i}"]), decision="deny")
        for i in range(80)
    ]
    compiled = compile_policy(pol.SafetyPolicy(rules=rules), words=1)
    assert not compiled.exact
    compiled2 = compile_policy(pol.SafetyPolicy(rules=rules), words=2)
    assert compiled2.exact


def test_legacy_tenant_policy_compiles():
    policy = pol.SafetyPolicy(
        tenants={
            "t1": pol.TenantPolicy(allow_topics=["job.echo"], deny_topics=["job.admin.*"]),
        }
    )
    compiled = compile_policy(policy)
    assert compiled.n_rules == 2  # deny + allow legacy rules
    enc = JobEncoder(compiled)
    batch = enc.encode([
        pol.PolicyInput(tenant="t1", topic="job.admin.users"),
        pol.PolicyInput(tenant="t1", topic="job.echo"),
        pol.PolicyInput(tenant="other", topic="job.admin.users"),
    ])
    got = first_match_reference(compiled, batch)
    assert compiled.rules[int(got[0])].decision == "deny"
    assert compiled.rules[int(got[1])].decision == "allow"
    assert int(got[2]) == -1


def test_empty_policy():
    compiled = compile_policy(None)
    enc = JobEncoder(compiled)
    batch = enc.encode([pol.PolicyInput(tenant="x", topic="job.y")])
    assert int(first_match_reference(compiled, batch)[0]) == -1


def test_synthetic_encoder_rows_are_exact_encodings():
    """SyntheticEncoder (the e2e bench's vectorized encoder) must emit rows
    that are exact JobEncoder encodings: every dimension of every job equals
    the LUT row built through the real encoder (or zero = absent)."""
    import torch

    from cordum_amd.ops.pipeline import SyntheticEncoder, make_synthetic_policy
    from cordum_amd.ops.policy_compile import (
        ALL_REQUIRES, DIM_RISK, DIM_TENANT, DIM_TOPIC, JobBatch, compile_policy,
    )

    compiled = compile_policy(make_synthetic_policy(256), words=1)
    assert compiled.exact
    enc = SyntheticEncoder(compiled, seed=3)
    J, W = 512, compiled.words
    out = JobBatch(
        torch.zeros((J, 7, W), dtype=torch.int64),
        torch.zeros((J, 2, W), dtype=torch.int64),
        torch.zeros((J,), dtype=torch.uint8),
        torch.zeros((J, 4, W), dtype=torch.int64),
        torch.zeros((J,), dtype=torch.uint8),
    )
    enc.fresh(out)
    tenant_rows = {int(r[0]) for r in enc.tenant_lut}
    topic_rows = {int(r[0]) for r in enc.topic_lut}
    risk_rows = {int(r[0]) for r in enc.risk_lut} | {0}
    req_rows = {int(r[0]) for r in enc.req_lut} | {0}
    for j in range(J):
        assert int(out.any_bits[j, DIM_TENANT, 0]) in tenant_rows
        assert int(out.any_bits[j, DIM_TOPIC, 0]) in topic_rows
        assert int(out.any_bits[j, DIM_RISK, 0]) in risk_rows
        assert int(out.all_bits[j, ALL_REQUIRES, 0]) in req_rows
    # fresh batches differ step to step
    out2 = JobBatch(out.any_bits.clone(), out.all_bits.clone(), out.secrets.clone(),
                    out.mcp_bits.clone(), out.mcp_used.clone())
    enc.fresh(out2)
    assert not torch.equal(out.any_bits, out2.any_bits)


def test_native_synthetic_encoder_oracle():
    """ext.synthetic_fresh (the fused host encoder the e2e bench uses) must
    match a pure-python splitmix64 replica row-for-row, be deterministic per
    (seed, step), and keep the torch path's distribution."""
    import pytest
    import torch

    from cordum_amd.ops import get_ext

    ext = get_ext(required=False)
    if ext is None or not hasattr(ext, "synthetic_fresh"):
        pytest.skip("HIP extension not built")
    from cordum_amd.ops.pipeline import DevicePipeline
    from cordum_amd.ops.policy_compile import DIM_RISK, DIM_TENANT, DIM_TOPIC

    pipe = DevicePipeline(device="cpu", batch_size=2048, n_local_workers=16,
                          n_rules=128, backend="ref", use_mfma=False)
    pipe.ensure_e2e()
    enc, host = pipe._e2e_enc, pipe._e2e_host

    enc.fresh_fast(host, 5, ext)
    a1 = host.any_bits.clone()
    enc.fresh_fast(host, 5, ext)
    assert torch.equal(host.any_bits, a1)  # deterministic per (seed, step)
    enc.fresh_fast(host, 6, ext)
    assert not torch.equal(host.any_bits, a1)

    M = (1 << 64) - 1

    def sm(x):
        x = (x + 0x9E3779B97F4A7C15) & M
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & M
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & M
        return x ^ (x >> 31)

    V = enc.tenant_lut.shape[0]
    enc.fresh_fast(host, 7, ext)
    base = sm((enc.seed * 0x5851F42D4C957F2D) & M ^ 7)

    def lemire(h32, v=V):  # index draw: multiply-shift reduction, not % V
        return ((h32 & 0xFFFFFFFF) * v) >> 32

    RISK_CUT = 1288490189  # 0.30 * 2^32
    for i in range(128):
        h0 = sm(base ^ i)
        h1 = sm(h0)
        h2 = sm(h1)
        assert torch.equal(host.any_bits[i, DIM_TENANT], enc.tenant_lut[lemire(h0)])
        assert torch.equal(host.any_bits[i, DIM_TOPIC], enc.topic_lut[lemire(h0 >> 32)])
        want = enc.risk_lut[lemire(h1)] if (h2 & 0xFFFFFFFF) < RISK_CUT \
            else torch.zeros_like(enc.risk_lut[0])
        assert torch.equal(host.any_bits[i, DIM_RISK], want)
    frac = float((host.any_bits[:, DIM_RISK].abs().sum(dim=1) != 0).float().mean())
    assert 0.2 < frac < 0.4


def test_alloc_batch_staging_layout():
    """The fused staging allocator must alias all five JobBatch fields into
    ONE buffer (so a single copy_ stages a batch) with 8-byte-aligned int64
    views and zero initial content; host and device layouts are the same
    code path, so asserting the CPU side pins the wire layout."""
    import torch

    from cordum_amd.ops.pipeline import alloc_batch_staging

    B, W = 64, 2
    jb, buf = alloc_batch_staging(B, W)
    assert buf.dtype == torch.uint8
    base = buf.data_ptr()
    end = base + buf.numel()
    for t, shape, dtype in [
        (jb.any_bits, (B, 7, W), torch.int64),
        (jb.all_bits, (B, 2, W), torch.int64),
        (jb.mcp_bits, (B, 4, W), torch.int64),
        (jb.secrets, (B,), torch.uint8),
        (jb.mcp_used, (B,), torch.uint8),
    ]:
        assert tuple(t.shape) == shape and t.dtype == dtype
        assert t.is_contiguous()
        assert base <= t.data_ptr() < end          # aliases the flat buffer
        if dtype == torch.int64:
            assert t.data_ptr() % 8 == 0
        assert int(t.abs().sum() if dtype == torch.int64 else t.sum()) == 0
    # writing a field is visible through the flat buffer (single-copy proof)
    jb.any_bits.fill_(-1)
    assert int((buf != 0).sum()) == B * 7 * W * 8
