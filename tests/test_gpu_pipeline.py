"""GPU pipeline tests: the fused (hipGraph-captured) tick must agree with the
host-side policy oracle on every batch, and state tables must be consistent."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fused_tick_counts_match_policy_oracle():
    from cordum_amd.ops.pipeline import DevicePipeline, SUCCEEDED, DENIED
    from cordum_amd.ops.policy_compile import first_match_reference

    pipe = DevicePipeline(
        device=torch.device("cuda:0"),
        batch_size=4096,
        n_local_workers=64,
        n_rules=512,
        n_batches=3,
        payload_words=32,
    )
    for i in range(4):
        st = pipe.tick()
        assert st.completed + st.denied + st.unrouted == 4096
        # independently recompute the expected deny count on the host
        jb = pipe.batches[(i) % 3]
        first = first_match_reference(pipe.compiled, _to_cpu(jb))
        dec = torch.where(
            first >= 0,
            pipe.compiled.decisions.to(torch.int32)[first.clamp(min=0).long()],
            torch.ones_like(first),
        )
        want_denied = int(((dec != 1) & (dec != 5)).sum())
        assert st.denied == want_denied
        assert st.completed == 4096 - want_denied  # all allowed jobs routable here
    # graph was actually captured and replayed
    assert pipe._graph is not None
    # job table consistency after the last tick
    states = pipe.states.cpu()
    assert int((states == SUCCEEDED).sum()) == st.completed
    assert int((states == DENIED).sum()) == st.denied


def _to_cpu(jb):
    from cordum_amd.ops.policy_compile import JobBatch

    return JobBatch(
        jb.any_bits.cpu(), jb.all_bits.cpu(), jb.secrets.cpu(),
        jb.mcp_bits.cpu(), jb.mcp_used.cpu(),
    )


def test_eager_and_fused_agree():
    from cordum_amd.ops.pipeline import DevicePipeline

    kw = dict(batch_size=2048, n_local_workers=32, n_rules=256, n_batches=2, payload_words=16, seed=3)
    fused = DevicePipeline(device=torch.device("cuda:0"), **kw)
    s1 = [fused.tick() for _ in range(2)]
    eager = DevicePipeline(device=torch.device("cuda:0"), **kw)
    eager._fused_capable = False  # force the eager path
    s2 = [eager._tick_eager() for _ in range(2)]
    for a, b in zip(s1, s2):
        assert (a.completed, a.denied, a.unrouted) == (b.completed, b.denied, b.unrouted)


def test_padded_graphed_path_matches_fused_world1():
    """Run the multi-rank (padded, segment-graphed) tick on one GPU with
    world=1 (collectives degrade to copies) and require identical per-tick
    counts to the fused single-GPU path — this validates the exact code the
    8-GPU bench runs, minus RCCL."""
    from cordum_amd.ops.pipeline import DevicePipeline

    kw = dict(batch_size=4096, n_local_workers=64, n_rules=512, n_batches=3,
              payload_words=16, seed=9)
    fused = DevicePipeline(device=torch.device("cuda:0"), **kw)
    s1 = [fused.tick() for _ in range(4)]

    padded = DevicePipeline(device=torch.device("cuda:0"), **kw)
    padded._fused_capable = False
    s2 = [padded._tick_padded() for _ in range(4)]
    assert padded._pad_graphs, "segment graphs must capture on GPU"
    for a, b in zip(s1, s2):
        assert (a.completed, a.denied, a.unrouted) == (b.completed, b.denied, b.unrouted)
    # state tables agree too
    assert torch.equal(fused.states.cpu(), padded.states.cpu())


def test_device_dlq_ring_collects_denied():
    from cordum_amd.ops.pipeline import DevicePipeline

    pipe = DevicePipeline(device=torch.device("cuda:0"), batch_size=4096,
                          n_local_workers=64, n_rules=512, n_batches=2, payload_words=8)
    total_denied = 0
    last_denied = 0
    for _ in range(3):
        st = pipe.tick()
        total_denied += st.denied
        last_denied = st.denied
    head = int(pipe.dlq_head.cpu()[0])
    assert head == total_denied
    ring = pipe.dlq_ring.cpu()
    entries = ring[ring >= 0]
    assert entries.numel() == min(total_denied, ring.numel())
    # the last tick's appends (ring[head-last : head]) point at slots that are
    # DENIED in the job table (earlier ticks' slots were recycled since —
    # states is per-tick working state, the ring is the durable record)
    states = pipe.states.cpu()
    rs = ring.numel()
    last_slots = [int(ring[p % rs]) for p in range(head - last_denied, head)]
    assert sorted(set(last_slots)) == sorted(last_slots)  # unique slots
    for slot in last_slots:
        assert states[slot] == 10  # DENIED


def test_tick_async_matches_tick_stats():
    from cordum_amd.ops.pipeline import DevicePipeline

    mk = lambda: DevicePipeline(device=torch.device("cuda:0"), batch_size=4096,
                                n_local_workers=64, n_rules=512, n_batches=2,
                                payload_words=8, seed=7)
    a = mk()
    sync_tot = [0, 0]
    for _ in range(6):
        st = a.tick()
        sync_tot[0] += st.completed
        sync_tot[1] += st.denied
    b = mk()
    b.tick()  # trigger capture (warmup resets the accumulator)
    b.reset_stats()
    for _ in range(6):
        b.tick_async()
    torch.cuda.synchronize()
    completed, denied = b.collect_stats()
    assert (completed, denied) == tuple(sync_tot)


def test_e2e_pipelined_deterministic_and_conserving():
    """The threaded/pipelined ingest window is deterministic (same seeds ->
    identical counts across runs), conserves every admitted job
    (completed + denied == steps * B), and really egresses per-batch
    results to host memory. Decision correctness itself is pinned by the
    kernel-vs-oracle suites; this checks the pipelining did not lose,
    duplicate, or fabricate work."""
    from cordum_amd.ops.pipeline import DevicePipeline

    def mk():
        return DevicePipeline(device="cuda:0", batch_size=4096,
                              n_local_workers=64, n_rules=256, n_batches=2,
                              backend="ext", use_mfma=False)

    steps = 12
    a = mk()
    ca, da, lats_a = a.e2e_run(steps)
    b = mk()
    cb, db, lats_b = b.e2e_run(steps)
    assert (ca, da) == (cb, db)
    assert ca + da == steps * 4096  # every job decided exactly once
    assert da > 0                   # the deny tail is real
    assert len(lats_a) == steps and all(l > 0 for l in lats_a)
    # per-batch result egress really happened: checksums are in host memory
    assert int(b._e2e_sums.abs().sum()) != 0
