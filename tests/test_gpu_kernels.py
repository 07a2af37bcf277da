"""GPU numerics tests: every HIP kernel vs its plain-torch reference
(ops/reference.py, ops/policy_compile.first_match_reference) and, for K1,
transitively vs the host policy evaluator oracle."""
import random

import pytest
import torch

pytestmark = pytest.mark.gpu

from cordum_amd.ops import get_ext
from cordum_amd.ops.policy_compile import JobEncoder, compile_policy, first_match_reference
from cordum_amd.ops.reference import (
    apply_transitions_ref,
    deadline_scan_ref,
    echo_execute_ref,
    least_loaded_pick_ref,
)
from cordum_amd.protocol.states import transition_lut
from tests.test_policy_compile import random_input, random_policy


@pytest.fixture(scope="module")
def ext():
    e = get_ext(required=True)
    return e


def dev():
    return torch.device("cuda:0")


def test_policy_first_match_matches_reference(ext):
    rng = random.Random(7)
    policy = random_policy(rng, n_rules=500, vocab=30)
    compiled = compile_policy(policy, words=1)
    assert compiled.exact
    enc = JobEncoder(compiled)
    inputs = [random_input(rng, vocab=30) for _ in range(2048)]
    batch = enc.encode(inputs)
    want = first_match_reference(compiled, batch)

    c = compiled.to(dev())
    b = batch.to(dev())
    got = ext.policy_first_match(
        c.any_masks, c.all_masks, c.secrets, c.mcp_masks, c.mcp_any,
        b.any_bits, b.all_bits, b.secrets, b.mcp_bits, b.mcp_used, 0,
    ).cpu()
    assert torch.equal(got, want)
    # transitive check against the host evaluator on a sample
    for j in rng.sample(range(len(inputs)), 64):
        d = policy.evaluate(inputs[j])
        idx = int(got[j])
        if idx < 0:
            assert d.rule_id == ""
        else:
            assert compiled.rules[idx].id == d.rule_id


def test_policy_first_match_large_rule_count(ext):
    """Config #4 shape: 100k-rule bundle, multi-word vocab."""
    rng = random.Random(11)
    policy = random_policy(rng, n_rules=2000, vocab=100)
    compiled = compile_policy(policy, words=2)
    assert compiled.exact
    enc = JobEncoder(compiled)
    inputs = [random_input(rng, vocab=100) for _ in range(512)]
    batch = enc.encode(inputs)
    want = first_match_reference(compiled, batch)
    c, b = compiled.to(dev()), batch.to(dev())
    got = ext.policy_first_match(
        c.any_masks, c.all_masks, c.secrets, c.mcp_masks, c.mcp_any,
        b.any_bits, b.all_bits, b.secrets, b.mcp_bits, b.mcp_used, 0,
    ).cpu()
    assert torch.equal(got, want)


def test_least_loaded_pick_matches_reference(ext):
    g = torch.Generator().manual_seed(3)
    NW, NJ = 1000, 4096
    w_pool = torch.randint(0, 8, (NW,), dtype=torch.int32, generator=g)
    w_active = torch.randint(0, 10, (NW,), dtype=torch.int32, generator=g)
    w_maxp = torch.randint(0, 12, (NW,), dtype=torch.int32, generator=g)
    w_cpu = torch.rand(NW, generator=g) * 100
    w_gpu = torch.rand(NW, generator=g) * 100
    w_labels = torch.randint(0, 16, (NW,), dtype=torch.int64, generator=g)
    j_poolmask = (1 << torch.randint(0, 8, (NJ,), generator=g)).to(torch.int64)
    j_labels = torch.where(torch.rand(NJ, generator=g) < 0.2,
                           torch.randint(0, 4, (NJ,), dtype=torch.int64, generator=g),
                           torch.zeros(NJ, dtype=torch.int64))
    want = least_loaded_pick_ref(w_pool, w_active, w_maxp, w_cpu, w_gpu, w_labels, j_poolmask, j_labels)
    d = dev()
    keys = ext.worker_precompute(w_pool.to(d), w_active.to(d), w_maxp.to(d), w_cpu.to(d), w_gpu.to(d))
    from cordum_amd.ops.reference import worker_precompute_ref

    assert torch.equal(keys.cpu(), worker_precompute_ref(w_pool, w_active, w_maxp, w_cpu, w_gpu))
    got = ext.least_loaded_pick(
        w_pool.to(d), keys, w_labels.to(d), j_poolmask.to(d), j_labels.to(d),
    ).cpu()
    assert torch.equal(got, want)


def test_apply_transitions_matches_reference(ext):
    g = torch.Generator().manual_seed(5)
    N, B = 10000, 4000
    states0 = torch.randint(0, 11, (N,), dtype=torch.uint8, generator=g)
    attempts0 = torch.zeros(N, dtype=torch.int32)
    deadlines0 = torch.randint(0, 1 << 40, (N,), dtype=torch.int64, generator=g)
    slots = torch.randperm(N, generator=g)[:B].to(torch.int32)  # unique slots
    to_states = torch.randint(0, 11, (B,), dtype=torch.uint8, generator=g)

    ref_states, ref_attempts, ref_deadlines = states0.clone(), attempts0.clone(), deadlines0.clone()
    want_ok = apply_transitions_ref(ref_states, ref_attempts, ref_deadlines, slots, to_states)

    d = dev()
    ext.set_transition_lut(torch.tensor(transition_lut(), dtype=torch.uint8).flatten())
    g_states, g_attempts, g_deadlines = states0.to(d), attempts0.to(d), deadlines0.to(d)
    got_ok = ext.apply_transitions(g_states, g_attempts, g_deadlines, slots.to(d), to_states.to(d)).cpu()
    assert torch.equal(got_ok, want_ok)
    assert torch.equal(g_states.cpu(), ref_states)
    assert torch.equal(g_attempts.cpu(), ref_attempts)
    assert torch.equal(g_deadlines.cpu(), ref_deadlines)


def test_deadline_scan_matches_reference(ext):
    g = torch.Generator().manual_seed(9)
    N = 50000
    states = torch.randint(0, 11, (N,), dtype=torch.uint8, generator=g)
    now = 1_000_000
    deadlines = torch.randint(0, 2_000_000, (N,), dtype=torch.int64, generator=g)
    updated = torch.randint(0, 2_000_000, (N,), dtype=torch.int64, generator=g)
    want = deadline_scan_ref(states, deadlines, updated, now, 500_000, 100_000)
    d = dev()
    slots, count = ext.deadline_scan(states.to(d), deadlines.to(d), updated.to(d),
                                     now, 500_000, 100_000, N)
    n = int(count.cpu()[0])
    got = slots[:n].cpu().sort().values
    assert torch.equal(got, want.sort().values)


def test_echo_execute_matches_reference(ext):
    g = torch.Generator().manual_seed(13)
    B, stride = 2048, 256
    ctx = torch.randint(-(1 << 31), (1 << 31) - 1, (B * stride,), dtype=torch.int32, generator=g)
    res_ref = torch.zeros_like(ctx)
    want = echo_execute_ref(ctx, res_ref, stride)
    d = dev()
    ctx_d = ctx.to(d)
    res_d = torch.zeros_like(ctx_d)
    got = ext.echo_execute(ctx_d, res_d, stride).cpu()
    assert torch.equal(res_d.cpu(), res_ref)
    assert torch.equal(got, want)


def test_echo_execute_indexed_matches_reference(ext):
    from cordum_amd.ops.reference import echo_execute_indexed_ref

    g = torch.Generator().manual_seed(21)
    N, stride = 4096, 64
    ctx = torch.randint(-(1 << 31), (1 << 31) - 1, (N * stride,), dtype=torch.int32, generator=g)
    slots = torch.randperm(N, generator=g)[:1500].to(torch.int32)
    res_ref = torch.zeros_like(ctx)
    sums_ref = torch.zeros(N, dtype=torch.int32)
    echo_execute_indexed_ref(ctx, slots, res_ref, sums_ref, stride)
    d = dev()
    ctx_d, res_d = ctx.to(d), torch.zeros_like(ctx).to(d)
    sums_d = torch.zeros(N, dtype=torch.int32, device=d)
    ext.echo_execute_indexed(ctx_d, slots.to(d), res_d, sums_d, stride)
    assert torch.equal(res_d.cpu(), res_ref)
    assert torch.equal(sums_d.cpu(), sums_ref)


def test_policy_mfma_variant_matches_reference(ext):
    """K1-MFMA must agree with the bitset kernel and the torch oracle
    (synthetic policies without MCP constraints — the variant's scope)."""
    from cordum_amd.ops.pipeline import make_synthetic_policy, encode_synthetic_jobs
    from cordum_amd.ops.policy_mfma import first_match_mfma, pack_jobs_mfma, pack_policy_mfma

    policy = make_synthetic_policy(777, vocab=40, deny_frac=0.05, seed=42)
    compiled = compile_policy(policy, words=1)
    assert compiled.exact
    jobs = encode_synthetic_jobs(compiled, 1000, vocab=40, seed=43)
    want = first_match_reference(compiled, jobs)

    d = dev()
    mp = pack_policy_mfma(compiled).to(d)
    a_pack, job_secrets = pack_jobs_mfma(jobs)
    got = first_match_mfma(ext, mp, a_pack.to(d), job_secrets.to(d), jobs.n_jobs).cpu()
    assert torch.equal(got, want)


def test_run_readiness_matches_reference(ext):
    from cordum_amd.ops.reference import run_readiness_ref

    g = torch.Generator().manual_seed(31)
    NR = 3000
    n_steps = torch.randint(1, 17, (NR,), dtype=torch.uint8, generator=g)
    step_state = torch.randint(0, 7, (NR, 64), dtype=torch.uint8, generator=g)
    deps_mask = torch.zeros(NR, 64, dtype=torch.int64)
    for r in range(NR):
        ns = int(n_steps[r])
        for s_i in range(ns):
            if s_i > 0 and torch.rand(1, generator=g).item() < 0.6:
                k = torch.randint(0, s_i, (min(3, s_i),), generator=g)
                for d_ in k.tolist():
                    deps_mask[r, s_i] |= 1 << d_
    run_active = (torch.rand(NR, generator=g) < 0.8).to(torch.uint8)
    want_ready, want_pairs = run_readiness_ref(step_state, deps_mask, n_steps, run_active)
    d = dev()
    ready, runs, steps, count = ext.run_readiness(
        step_state.to(d), deps_mask.to(d), n_steps.to(d), run_active.to(d), NR * 64)
    assert torch.equal(ready.cpu(), want_ready)
    n = int(count.cpu()[0])
    got_pairs = sorted(zip(runs[:n].cpu().tolist(), steps[:n].cpu().tolist()))
    assert got_pairs == sorted(want_pairs)


def test_padded_dispatch_kernels_match_reference(ext):
    """pack_by_dest / gather_payload_padded / echo_padded /
    apply_transitions_padded / load_feedback_padded vs their CPU oracles."""
    from cordum_amd.ops import reference as ref

    g = torch.Generator().manual_seed(77)
    B, world, nwl, stride = 2048, 4, 50, 16
    n = 1500
    routable_count = torch.tensor([n], dtype=torch.int32)
    routable_slots = torch.randperm(B, generator=g)[:n].to(torch.int32)
    routable_widx = torch.randint(0, world * nwl, (n,), dtype=torch.int32, generator=g)
    payload = torch.randint(-(1 << 31), (1 << 31) - 1, (B * stride,), dtype=torch.int32, generator=g)

    def rq_bufs(dev_=None):
        kw = {"device": dev_} if dev_ is not None else {}
        return (torch.zeros(B, dtype=torch.int32, **kw),
                torch.zeros(B, dtype=torch.int32, **kw),
                torch.zeros(B, dtype=torch.int32, **kw),
                torch.zeros(1, dtype=torch.int32, **kw),
                torch.zeros(1, dtype=torch.int64, **kw),
                torch.zeros(8, dtype=torch.int32, **kw),
                torch.zeros(1, dtype=torch.int32, **kw))

    # references
    ss_r = torch.zeros(world * B, dtype=torch.int32)
    sw_r = torch.zeros(world * B, dtype=torch.int32)
    sc_r = torch.zeros(world, dtype=torch.int32)
    rq_r = rq_bufs()
    ref.pack_by_dest_ref(routable_slots, routable_widx, routable_count, ss_r, sw_r, sc_r, nwl, B,
                         *rq_r)
    rq_prev_payload = torch.zeros(B * stride, dtype=torch.int32)
    sp_r = torch.zeros(world * B * stride, dtype=torch.int32)
    ref.gather_payload_padded_ref(payload, rq_prev_payload, ss_r, sc_r, sp_r, stride, B, world)
    ra_r = torch.zeros_like(sp_r)
    sums_r = torch.zeros(world * B, dtype=torch.int32)
    ref.echo_padded_ref(sp_r, sc_r, ra_r, sums_r, stride, B, world)
    states_r = torch.full((B,), 4, dtype=torch.uint8)  # DISPATCHED
    att_r = torch.zeros(B, dtype=torch.int32)
    dl_r = torch.zeros(B, dtype=torch.int64)
    ref.apply_transitions_padded_ref(states_r, att_r, dl_r, ss_r, sc_r, 5, B, world)
    wal_r = torch.zeros(nwl, dtype=torch.int32)
    ref.load_feedback_padded_ref(sw_r, sc_r, wal_r, B, world)

    # device
    d = dev()
    ss = torch.zeros(world * B, dtype=torch.int32, device=d)
    sw = torch.zeros(world * B, dtype=torch.int32, device=d)
    sc = torch.zeros(world, dtype=torch.int32, device=d)
    rq_d = rq_bufs(d)
    ext.pack_by_dest(routable_slots.to(d), routable_widx.to(d), routable_count.to(d),
                     ss, sw, sc, nwl, B, n, *rq_d)
    assert torch.equal(sc.cpu(), sc_r)
    assert int(rq_d[3].cpu()[0]) == 0 and int(rq_d[4].cpu()[0]) == 0  # cap=B: no overflow
    # per-dest sets must match (order within a segment may differ)
    for r in range(world):
        k = int(sc_r[r])
        a = sorted(zip(ss[r * B:r * B + k].cpu().tolist(), sw[r * B:r * B + k].cpu().tolist()))
        b = sorted(zip(ss_r[r * B:r * B + k].tolist(), sw_r[r * B:r * B + k].tolist()))
        assert a == b
    # use the REFERENCE packing on device for the order-dependent stages
    ss_d, sw_d, sc_d = ss_r.to(d), sw_r.to(d), sc_r.to(d)
    sp = torch.zeros(world * B * stride, dtype=torch.int32, device=d)
    ext.gather_payload_padded(payload.to(d), rq_prev_payload.to(d), ss_d, sc_d, sp, stride, B, world)
    assert torch.equal(sp.cpu(), sp_r)
    ra = torch.zeros_like(sp)
    sums = torch.zeros(world * B, dtype=torch.int32, device=d)
    ext.echo_padded(sp, sc_d, ra, sums, stride, B, world)
    assert torch.equal(ra.cpu(), ra_r)
    assert torch.equal(sums.cpu(), sums_r)
    states = torch.full((B,), 4, dtype=torch.uint8, device=d)
    att = torch.zeros(B, dtype=torch.int32, device=d)
    dl = torch.zeros(B, dtype=torch.int64, device=d)
    ext.set_transition_lut(torch.tensor(transition_lut(), dtype=torch.uint8).flatten())
    ext.apply_transitions_padded(states, att, dl, ss_d, sc_d, 5, B, world)
    assert torch.equal(states.cpu(), states_r)
    wal = torch.zeros(nwl, dtype=torch.int32, device=d)
    ext.load_feedback_padded(sw_d, sc_d, wal, B, world)
    assert torch.equal(wal.cpu(), wal_r)


def test_requeue_kernels_match_reference(ext):
    """pack_by_dest overflow parking + pack_requeue redelivery +
    materialize_rq_payload vs the CPU oracles, with a forced-small cap."""
    from cordum_amd.ops import reference as ref

    g = torch.Generator().manual_seed(99)
    B, world, nwl, stride, cap = 512, 4, 8, 8, 16
    n = 400
    routable_count = torch.tensor([n], dtype=torch.int32)
    routable_slots = torch.randperm(B, generator=g)[:n].to(torch.int32)
    routable_widx = torch.randint(0, world * nwl, (n,), dtype=torch.int32, generator=g)
    payload = torch.randint(-(1 << 31), (1 << 31) - 1, (B * stride,),
                            dtype=torch.int32, generator=g)
    prev_payload = torch.randint(-(1 << 31), (1 << 31) - 1, (B * stride,),
                                 dtype=torch.int32, generator=g)
    prev_widx = torch.randint(0, world * nwl, (B,), dtype=torch.int32, generator=g)
    prev_att = torch.randint(1, 5, (B,), dtype=torch.int32, generator=g)
    prev_count = torch.tensor([60], dtype=torch.int32)

    def mk(dev_=None):
        kw = {"device": dev_} if dev_ is not None else {}
        return {
            "ss": torch.zeros(world * cap, dtype=torch.int32, **kw),
            "sw": torch.zeros(world * cap, dtype=torch.int32, **kw),
            "sc": torch.zeros(world, dtype=torch.int32, **kw),
            "rq_src": torch.zeros(B, dtype=torch.int32, **kw),
            "rq_widx": torch.zeros(B, dtype=torch.int32, **kw),
            "rq_att": torch.zeros(B, dtype=torch.int32, **kw),
            "rq_count": torch.zeros(1, dtype=torch.int32, **kw),
            "rq_dead": torch.zeros(1, dtype=torch.int64, **kw),
            "rq_payload": torch.zeros(B * stride, dtype=torch.int32, **kw),
            "dead_src": torch.zeros(B, dtype=torch.int32, **kw),
            "dead_count": torch.zeros(1, dtype=torch.int32, **kw),
        }

    r = mk()
    ref.pack_requeue_ref(prev_widx, prev_att, prev_count, r["ss"], r["sw"], r["sc"],
                         nwl, cap, r["rq_src"], r["rq_widx"], r["rq_att"],
                         r["rq_count"], r["rq_dead"], r["dead_src"], r["dead_count"])
    ref.pack_by_dest_ref(routable_slots, routable_widx, routable_count,
                         r["ss"], r["sw"], r["sc"], nwl, cap,
                         r["rq_src"], r["rq_widx"], r["rq_att"],
                         r["rq_count"], r["rq_dead"], r["dead_src"], r["dead_count"])
    ref.materialize_rq_payload_ref(payload, prev_payload, r["rq_src"],
                                   r["rq_count"], r["rq_payload"], stride)

    d = dev()
    v = mk(d)
    ext.pack_requeue(prev_widx.to(d), prev_att.to(d), prev_count.to(d),
                     v["ss"], v["sw"], v["sc"], nwl, cap,
                     v["rq_src"], v["rq_widx"], v["rq_att"],
                     v["rq_count"], v["rq_dead"], v["dead_src"], v["dead_count"])
    ext.pack_by_dest(routable_slots.to(d), routable_widx.to(d), routable_count.to(d),
                     v["ss"], v["sw"], v["sc"], nwl, cap, n,
                     v["rq_src"], v["rq_widx"], v["rq_att"],
                     v["rq_count"], v["rq_dead"], v["dead_src"], v["dead_count"])
    ext.materialize_rq_payload(payload.to(d), prev_payload.to(d), v["rq_src"],
                               v["rq_count"], v["rq_payload"], stride)

    # Counts agree exactly. WHICH entries win send slots vs park under
    # contention is atomic-order dependent, so contents are compared as the
    # order-invariant PARTITION UNION: every input identity lands in exactly
    # one of {packed, parked, dead} and the union must match the reference's.
    assert torch.equal(v["sc"].cpu().clamp(max=cap), r["sc"].clamp(max=cap))
    assert int(v["rq_count"].cpu()[0]) == int(r["rq_count"][0])
    assert int(v["rq_dead"].cpu()[0]) == int(r["rq_dead"][0])
    assert int(v["dead_count"].cpu()[0]) == int(r["dead_count"][0])

    def identities(b):
        ids = []
        for q in range(world):
            k = min(int(b["sc"][q] if b is r else b["sc"].cpu()[q]), cap)
            seg = (b["ss"] if b is r else b["ss"].cpu())[q * cap:q * cap + k]
            ids += [("packed", int(x)) for x in seg.tolist()]
        nq_ = min(int((b["rq_count"] if b is r else b["rq_count"].cpu())[0]), B)
        ids += [("parked", int(x)) for x in
                (b["rq_src"] if b is r else b["rq_src"].cpu())[:nq_].tolist()]
        nd_ = min(int((b["dead_count"] if b is r else b["dead_count"].cpu())[0]), B)
        ids += [("dead", int(x)) for x in
                (b["dead_src"] if b is r else b["dead_src"].cpu())[:nd_].tolist()]
        return sorted(x for _, x in ids)

    assert identities(v) == identities(r)
    # parked attempts are identity-determined: flagged -> prev+1, fresh -> 1
    nq = min(int(r["rq_count"][0]), B)
    for j in range(nq):
        s = int(v["rq_src"].cpu()[j])
        att = int(v["rq_att"].cpu()[j])
        assert att == (int(prev_att[-1 - s]) + 1 if s < 0 else 1)
    # materialized payload rows: each device ring entry's payload matches the
    # row its (device-order) src points at
    pl = payload.view(B, stride)
    prev = prev_payload.view(B, stride)
    vp = v["rq_payload"].cpu().view(B, stride)
    vsrc = v["rq_src"].cpu()
    for j in range(nq):
        s = int(vsrc[j])
        want_row = pl[s] if s >= 0 else prev[-1 - s]
        assert torch.equal(vp[j], want_row)


def test_fused_tick_kernels_match_refops(ext):
    """Direct unit checks of the launch-fusion kernels (policy_gate_full,
    apply_transitions_chain_dyn, begin_tick, compact_routable_spread,
    accumulate_counts) against the CPU _RefOps oracles on random data."""
    from cordum_amd.ops.pipeline import _RefOps

    torch.manual_seed(5)
    d = dev()
    ref = _RefOps()
    lut = torch.tensor(transition_lut(), dtype=torch.uint8).flatten()
    ref.set_transition_lut(lut)
    ext.set_transition_lut(lut)
    B, R = 2048, 96

    # -- policy_gate_full ----------------------------------------------------
    first = torch.where(torch.rand(B) < 0.7,
                        torch.randint(0, R, (B,), dtype=torch.int32),
                        torch.tensor(2**31 - 1, dtype=torch.int32))
    decisions = torch.randint(1, 6, (R,), dtype=torch.int8)

    def gate_state():
        states = torch.ones(B, dtype=torch.uint8)  # PENDING
        deadlines = torch.randint(1, 2**40, (B,), dtype=torch.int64)
        out_dec = torch.zeros(B, dtype=torch.int8)
        ds = torch.zeros(B, dtype=torch.int32)
        als = torch.zeros(B, dtype=torch.int32)
        cnt = torch.zeros(4, dtype=torch.int32)
        ring = torch.full((4096,), -1, dtype=torch.int32)  # > B: no wrap in-test
        head = torch.zeros(1, dtype=torch.int32)
        return states, deadlines, out_dec, ds, als, cnt, ring, head

    torch.manual_seed(11)
    s1 = gate_state()
    ref.policy_gate_full(first, decisions, s1[2], s1[3], s1[5][0:1], s1[4], s1[5][1:2],
                         s1[0], s1[1], s1[6], s1[7])
    torch.manual_seed(11)
    s2 = [t.to(d) for t in gate_state()]
    ext.policy_gate_full(first.to(d), decisions.to(d), s2[2], s2[3], s2[5][0:1],
                         s2[4], s2[5][1:2], s2[0], s2[1], s2[6], s2[7])
    assert torch.equal(s1[0], s2[0].cpu())          # states (DENIED applied)
    assert torch.equal(s1[1], s2[1].cpu())          # deadlines cleared
    assert torch.equal(s1[2], s2[2].cpu())          # decisions
    assert int(s1[5][0]) == int(s2[5][0].cpu())     # denied count
    assert int(s1[5][1]) == int(s2[5][1].cpu())     # allowed count
    assert int(s1[7][0]) == int(s2[7][0].cpu())     # dlq head
    # ring contents as sets (append order differs across waves)
    n = int(s1[7][0])
    assert set(s1[6][:n].tolist()) == set(s2[6].cpu()[s2[6].cpu() >= 0].tolist())
    # compacted slot sets match
    assert set(s1[3][: int(s1[5][0])].tolist()) == set(s2[3].cpu()[: int(s1[5][0])].tolist())
    assert set(s1[4][: int(s1[5][1])].tolist()) == set(s2[4].cpu()[: int(s1[5][1])].tolist())

    # -- apply_transitions_chain_dyn ----------------------------------------
    slots = torch.randperm(B, dtype=torch.int64)[: B // 2].to(torch.int32)
    count = torch.tensor([slots.numel()], dtype=torch.int32)
    chain = [3, 4, 5, 6]  # SCHEDULED..SUCCEEDED

    def chain_state():
        states = torch.ones(B, dtype=torch.uint8)
        states[::7] = 10  # some terminal slots: chain must not move them
        attempts = torch.zeros(B, dtype=torch.int32)
        deadlines = torch.randint(1, 2**40, (B,), dtype=torch.int64)
        extra = torch.randint(1, 100, (64,), dtype=torch.int32)
        return states, attempts, deadlines, extra

    torch.manual_seed(13)
    c1 = chain_state()
    ref.apply_transitions_chain_dyn(c1[0], c1[1], c1[2], slots, count, chain, c1[3], B)
    torch.manual_seed(13)
    c2 = [t.to(d) for t in chain_state()]
    ext.apply_transitions_chain_dyn(c2[0], c2[1], c2[2], slots.to(d), count.to(d),
                                    chain, c2[3], B)
    for i in range(4):
        assert torch.equal(c1[i], c2[i].cpu()), f"chain tensor {i}"

    # -- begin_tick + accumulate_counts -------------------------------------
    st = torch.zeros(B, dtype=torch.uint8, device=d)
    cnts = torch.tensor([3, 5, 7, 9], dtype=torch.int32, device=d)
    ext.begin_tick(st, cnts)
    assert int(st.sum().cpu()) == B and int(cnts.abs().sum().cpu()) == 0
    acc = torch.zeros(4, dtype=torch.int64, device=d)
    src = torch.tensor([1, 2, 3, 4], dtype=torch.int32, device=d)
    ext.accumulate_counts(src, acc)
    ext.accumulate_counts(src, acc)
    assert acc.cpu().tolist() == [2, 4, 6, 8]


def test_pack_jobs_mfma_device_matches_host(ext):
    """The device A-fragment packing must equal policy_mfma.pack_jobs_mfma
    byte-for-byte (same fragment layout the matrix cores consume)."""
    from cordum_amd.ops.pipeline import encode_synthetic_jobs, make_synthetic_policy
    from cordum_amd.ops.policy_compile import compile_policy
    from cordum_amd.ops.policy_mfma import N_DIMS, pack_jobs_mfma

    policy = make_synthetic_policy(300, vocab=40, seed=3)
    compiled = compile_policy(policy, words=1)
    assert compiled.exact
    jb = encode_synthetic_jobs(compiled, 1000, seed=17)
    want, _ = pack_jobs_mfma(jb)

    d = dev()
    jb_d = jb.to(d)
    Jt = (1000 + 15) // 16
    got = torch.zeros(Jt, N_DIMS, 64, 16, dtype=torch.int8, device=d)
    ext.pack_jobs_mfma_dev(jb_d.any_bits, jb_d.all_bits, got)
    assert torch.equal(got.cpu(), want)


def test_e2e_mfma_device_pack_counts_match_bitset(ext):
    """The e2e window with device-packed MFMA K1 must decide exactly what
    the bitset K1 decides on identical staged batches."""
    from cordum_amd.ops.pipeline import DevicePipeline

    def run(**kw):
        p = DevicePipeline(device="cuda:0", batch_size=4096, n_local_workers=64,
                           n_rules=512, n_batches=2, backend="ext", **kw)
        return p.e2e_run(8)

    c1, d1, _ = run(use_mfma=False)
    c2, d2, _ = run(mfma_pack_on_device=True)
    assert (c1, d1) == (c2, d2)
    assert d1 > 0
