"""GPU tests for the device workflow engine (K3-WF kernels): the HIP tick
must produce the same terminal run states, child accounting and retry
counts as the CPU reference backend on identical DAGs."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from cordum_amd.ops.wf_pipeline import (
    DagSpec,
    StepSpec,
    WFK_APPROVAL,
    WFK_CONDITION,
    WFK_DELAY,
    WFK_FOR_EACH,
    WFK_WORKER,
    WorkflowPipeline,
)


def mk(backend, device, dags, **kw):
    kw.setdefault("n_local_workers", 32)
    kw.setdefault("payload_words", 8)
    return WorkflowPipeline(device=device, backend=backend, dags=dags, **kw)


def run_both(dags, **kw):
    ref = mk("ref", "cpu", dags, **kw)
    st_r = ref.run_wave()
    ext = mk("ext", "cuda:0", dags, **kw)
    st_e = ext.run_wave()
    return ref, st_r, ext, st_e


def test_config3_wave_matches_reference():
    dags = [DagSpec.fanout_approval(64) for _ in range(16)]
    ref, st_r, ext, st_e = run_both(dags)
    assert (st_e.runs_succeeded, st_e.runs_failed) == (st_r.runs_succeeded, st_r.runs_failed)
    assert torch.equal(ext.run_state.cpu(), ref.run_state)
    assert torch.equal(ext.children_done.cpu(), ref.children_done)
    assert torch.equal(ext.step_state.cpu(), ref.step_state)


def test_retry_wave_matches_reference():
    """Identical deterministic failure injection on both backends -> the
    retry counters and terminal states agree exactly (ticks in lockstep:
    both run the same fixed tick count)."""
    dags = [DagSpec(steps=[StepSpec(WFK_FOR_EACH, fanout=32),
                           StepSpec(WFK_WORKER, deps=[0])])
            for _ in range(8)]
    kw = dict(fail_ppt=150, max_retries=8)
    ref = mk("ref", "cpu", dags, **kw)
    ext = mk("ext", "cuda:0", dags, **kw)
    ref.reset_runs()
    ext.reset_runs()
    for _ in range(64):
        ref.tick()
        ext.tick()
    assert torch.equal(ext.run_state.cpu(), ref.run_state)
    assert torch.equal(ext.step_attempts.cpu(), ref.step_attempts)
    assert torch.equal(ext.children_done.cpu(), ref.children_done)
    assert ext.counts() == ref.counts()


def test_mixed_kinds_match_reference():
    import random

    rng = random.Random(7)
    dags = []
    for _ in range(32):
        n = rng.randint(1, 8)
        steps = []
        for s in range(n):
            deps = [d for d in range(s) if rng.random() < 0.4]
            kind = rng.choice([WFK_WORKER, WFK_FOR_EACH, WFK_APPROVAL,
                               WFK_CONDITION, WFK_DELAY])
            steps.append(StepSpec(kind, deps=deps,
                                  fanout=rng.randint(1, 17) if kind == WFK_FOR_EACH else 1,
                                  delay_ticks=rng.randint(0, 4),
                                  cond=rng.random() < 0.5))
        dags.append(DagSpec(steps=steps))
    ref, st_r, ext, st_e = run_both(dags)
    assert torch.equal(ext.run_state.cpu(), ref.run_state)
    assert torch.equal(ext.step_state.cpu(), ref.step_state)


def test_wave_throughput_smoke():
    """Config #3 at bench shape on one GPU: a full wave completes and the
    native extension is the execution engine."""
    import time

    dags = [DagSpec.fanout_approval(256) for _ in range(64)]
    pipe = mk("ext", "cuda:0", dags)
    pipe.run_wave()  # warm
    t0 = time.perf_counter()
    st = pipe.run_wave()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert st.runs_succeeded == 64
    assert dt < 5.0
