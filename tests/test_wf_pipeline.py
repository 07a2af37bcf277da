"""Device workflow-engine tick (ops/wf_pipeline.py) on the CPU reference
backend: config #3 semantics (fan-out + approval), retry waves, dead-letter
accounting, and equivalence with the host workflow engine
(cordum_amd/workflow/engine.py — itself the port of core/workflow/engine.go)
on the same DAGs."""
import json
import os

import pytest
import torch
import torch.multiprocessing as mp

from cordum_amd.ops.wf_pipeline import (
    DagSpec,
    StepSpec,
    WFK_APPROVAL,
    WFK_CONDITION,
    WFK_DELAY,
    WFK_FOR_EACH,
    WFK_WORKER,
    WFS_FAILED,
    WFS_SUCCEEDED,
    WorkflowPipeline,
)

pytestmark = pytest.mark.timeout(300)


def mk_pipe(dags, **kw):
    kw.setdefault("device", "cpu")
    kw.setdefault("backend", "ref")
    kw.setdefault("n_local_workers", 16)
    kw.setdefault("payload_words", 4)
    return WorkflowPipeline(dags=dags, **kw)


def test_config3_fanout_approval_wave_succeeds():
    """Config #3 DAG (seed -> 1->N fan-out -> approval -> final) completes
    with exact child accounting."""
    NRUNS, FAN = 8, 32
    pipe = mk_pipe([DagSpec.fanout_approval(FAN) for _ in range(NRUNS)])
    st = pipe.run_wave()
    assert st.runs_succeeded == NRUNS and st.runs_failed == 0
    done = pipe.children_done.view(-1, 64)
    for r in range(NRUNS):
        assert int(done[r, 0]) == 1          # seed
        assert int(done[r, 1]) == FAN        # every fan-out child exactly once
        assert int(done[r, 3]) == 1          # final
    assert int(pipe.rq_dead[0]) == 0
    # approval actually held: the wave needs the host grant hop
    assert st.ticks >= 4


def test_mixed_dags_condition_delay_skip():
    """CONDITION false -> SKIPPED satisfies deps; DELAY gates by ticks."""
    dag = DagSpec(steps=[
        StepSpec(WFK_CONDITION, cond=False),
        StepSpec(WFK_DELAY, deps=[0], delay_ticks=5),
        StepSpec(WFK_WORKER, deps=[1]),
    ])
    pipe = mk_pipe([dag] * 4)
    st = pipe.run_wave()
    assert st.runs_succeeded == 4
    assert st.ticks >= 5  # the delay gate really held
    states = pipe.step_state.view(-1, 64)
    assert int(states[0, 0]) == 5  # WFS_SKIPPED
    assert int(states[0, 1]) == WFS_SUCCEEDED
    assert int(states[0, 2]) == WFS_SUCCEEDED


def test_retry_waves_eventually_succeed():
    """Failure injection + generous retries: every failed child re-runs
    until the step aggregates fanout successes (per-child retry, not
    whole-step)."""
    NRUNS, FAN = 6, 16
    dags = [DagSpec(steps=[StepSpec(WFK_FOR_EACH, fanout=FAN)])
            for _ in range(NRUNS)]
    pipe = mk_pipe(dags, fail_ppt=200, max_retries=10)
    st = pipe.run_wave()
    assert st.runs_succeeded == NRUNS, (st, pipe.counts())
    done = pipe.children_done.view(-1, 64)
    att = pipe.step_attempts.view(-1, 64)
    assert all(int(done[r, 0]) == FAN for r in range(NRUNS))
    assert any(int(att[r, 0]) > 0 for r in range(NRUNS))  # retries happened


def test_max_retries_exhaustion_fails_run():
    dags = [DagSpec(steps=[StepSpec(WFK_WORKER),
                           StepSpec(WFK_WORKER, deps=[0])])] * 3
    pipe = mk_pipe(list(dags), fail_ppt=1000, max_retries=2)  # always fail
    st = pipe.run_wave()
    assert st.runs_failed == 3 and st.runs_succeeded == 0
    att = pipe.step_attempts.view(-1, 64)
    states = pipe.step_state.view(-1, 64)
    for r in range(3):
        assert int(states[r, 0]) == WFS_FAILED
        assert int(att[r, 0]) == 2           # retried exactly max_retries times
        assert int(states[r, 1]) == 0        # downstream blocked, never ran


def test_approval_rejection_fails_run():
    pipe = mk_pipe([DagSpec.fanout_approval(8)] * 2, approval_verdict=0)
    st = pipe.run_wave()
    assert st.runs_failed == 2 and st.runs_succeeded == 0


def test_arena_backpressure_still_completes():
    """Child arena smaller than one wave's fan-out: all-or-nothing
    reservation spreads emission over ticks, nothing is lost."""
    NRUNS, FAN = 6, 64
    pipe = mk_pipe([DagSpec(steps=[StepSpec(WFK_FOR_EACH, fanout=FAN)])
                    for _ in range(NRUNS)], child_cap=128)
    st = pipe.run_wave()
    assert st.runs_succeeded == NRUNS
    done = pipe.children_done.view(-1, 64)
    assert all(int(done[r, 0]) == FAN for r in range(NRUNS))


def test_pad_overflow_requeue_still_completes():
    """Destination capacity forced tiny: children park in the requeue ring
    and redeliver; steps complete once every child finally lands."""
    NRUNS, FAN = 4, 32
    pipe = mk_pipe([DagSpec(steps=[StepSpec(WFK_FOR_EACH, fanout=FAN)])
                    for _ in range(NRUNS)], pad_cap=16, max_retries=0)
    st = pipe.run_wave(max_ticks=512)
    assert st.runs_succeeded + st.runs_failed == NRUNS
    done = pipe.children_done.view(-1, 64)
    fail_total = int(pipe.children_fail.view(-1).sum())
    total = sum(int(done[r, 0]) for r in range(NRUNS)) + fail_total + int(pipe.rq_dead[0])
    # conservation across redelivery: every emitted child is eventually
    # applied or counted dead
    assert total >= NRUNS * FAN


def _host_oracle_run(dag: DagSpec, fanout_items: int):
    """Run the equivalent workflow through the HOST engine with an
    auto-succeeding worker; return the final run status string."""
    from cordum_amd.bus import LoopbackBus
    from cordum_amd.protocol import subjects as subj
    from cordum_amd.protocol.capv2 import JobResult, JobStatus
    from cordum_amd.store import MemoryStore
    from cordum_amd.utils.clock import ManualClock
    from cordum_amd.workflow import Engine, Step, Workflow, WorkflowRun, WorkflowStore

    clock = ManualClock()
    bus = LoopbackBus(clock=clock)
    store = WorkflowStore(clock=clock)
    memory = MemoryStore(clock=clock)
    engine = Engine(store, bus, memory=memory, clock=clock)
    pending = []
    bus.subscribe(subj.SUBJECT_SUBMIT, lambda s, p: pending.append(p.job_request))

    steps = {}
    for i, spec in enumerate(dag.steps):
        sid = f"s{i}"
        deps = [f"s{d}" for d in spec.deps]
        if spec.kind == WFK_WORKER:
            steps[sid] = {"type": "worker", "topic": "job.t", "depends_on": deps}
        elif spec.kind == WFK_FOR_EACH:
            steps[sid] = {"type": "worker", "topic": "job.t", "depends_on": deps,
                          "for_each": "${input.items}"}
            if spec.max_parallel:
                steps[sid]["max_parallel"] = spec.max_parallel
        elif spec.kind == WFK_APPROVAL:
            steps[sid] = {"type": "approval", "depends_on": deps}
        elif spec.kind == WFK_DELAY:
            steps[sid] = {"type": "delay", "delay_sec": spec.delay_ticks,
                          "depends_on": deps}
        else:
            raise AssertionError("oracle comparison covers worker/for_each/approval/delay")
    wf = Workflow(id="wf", org_id="org",
                  steps={sid: Step.from_dict(sid, sd) for sid, sd in steps.items()})
    store.put_workflow(wf)
    run = WorkflowRun(id="r1", workflow_id="wf", org_id="org",
                      input={"items": list(range(fanout_items))})
    store.create_run(run)
    engine.start_run("wf", "r1")
    for _ in range(10000):
        if pending:
            req = pending.pop(0)
            ptr = memory.put_result(req.job_id, json.dumps({"ok": 1}).encode())
            engine.handle_job_result(JobResult(job_id=req.job_id,
                                               status=JobStatus.SUCCEEDED,
                                               result_ptr=ptr))
            continue
        r = store.get_run("r1")
        if r.status in ("succeeded", "failed", "cancelled", "timed_out"):
            return r.status
        if r.status == "waiting":
            for sid, sr in list(r.steps.items()):
                if sr.status == "waiting":
                    engine.approve_step("r1", sid, True)
            continue
        clock.advance(30)
        if engine.pump_timers() == 0 and not pending:
            r = store.get_run("r1")
            if r.status in ("succeeded", "failed"):
                return r.status
    raise AssertionError("host oracle did not converge")


def test_device_tick_matches_host_engine_on_random_dags():
    """fail_ppt=0 equivalence: the device tick and the host engine agree on
    the terminal status of random worker/for_each/approval/delay DAGs."""
    import random

    rng = random.Random(42)
    dags = []
    for _ in range(12):
        n = rng.randint(1, 6)
        steps = []
        for s in range(n):
            deps = [d for d in range(s) if rng.random() < 0.5]
            kind = rng.choice([WFK_WORKER, WFK_FOR_EACH, WFK_APPROVAL, WFK_DELAY])
            steps.append(StepSpec(kind, deps=deps,
                                  fanout=rng.randint(1, 9) if kind == WFK_FOR_EACH else 1,
                                  max_parallel=rng.choice([0, 0, 2, 3])
                                  if kind == WFK_FOR_EACH else 0,
                                  delay_ticks=rng.randint(0, 3)))
        dags.append(DagSpec(steps=steps))
    pipe = mk_pipe(dags)
    st = pipe.run_wave()
    assert st.runs_succeeded + st.runs_failed == len(dags)
    run_state = pipe.run_state.cpu()
    for i, dag in enumerate(dags):
        fan = next((s.fanout for s in dag.steps if s.kind == WFK_FOR_EACH), 1)
        want = _host_oracle_run(dag, fan)
        got = "succeeded" if int(run_state[i]) == WFS_SUCCEEDED else "failed"
        assert got == want, (i, got, want)


def _wf_dist_worker(rank, world, port, result_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    NRUNS, FAN = 4, 16
    pipe = mk_pipe([DagSpec.fanout_approval(FAN) for _ in range(NRUNS)],
                   world_size=world, rank=rank)
    remote = 0
    pipe.reset_runs()
    for _ in range(256):  # lockstep: every rank ticks the same count
        pipe.tick()
        # children received from OTHER ranks this tick (the xGMI traffic)
        rc = pipe.pad_recv_cnt.cpu()
        remote += int(rc.sum()) - int(rc[rank])
        done = torch.tensor([0 if pipe.active() == 0 else 1], dtype=torch.int64)
        dist.all_reduce(done)
        if int(done.item()) == 0:
            break
    ok, fail = pipe.counts()
    out = {"ok": ok, "fail": fail, "remote_work": remote}
    with open(os.path.join(result_dir, f"rank{rank}.json"), "w") as f:
        json.dump(out, f)
    dist.barrier()
    dist.destroy_process_group()


def test_two_rank_workflow_fanout(tmp_path):
    """Config #3 across 2 ranks: children cross the exchange, every run on
    every rank completes."""
    mp.spawn(_wf_dist_worker, args=(2, 29741, str(tmp_path)), nprocs=2, join=True)
    for r in range(2):
        out = json.load(open(tmp_path / f"rank{r}.json"))
        assert out["ok"] == 4 and out["fail"] == 0
        assert out["remote_work"] > 0


def test_continuous_soak_reconciles():
    """Config #5 mechanics in miniature: continuous re-admission holds the
    table full; injected failures retry with backoff; injected LOST results
    are recovered by the K4-WF timeout scan; failed runs drain to the host
    DLQ; device counters reconcile exactly against host bookkeeping."""
    from cordum_amd.store.dlq_store import DLQStore

    NR = 32
    dag = DagSpec(steps=[
        StepSpec(WFK_WORKER),
        StepSpec(WFK_FOR_EACH, deps=[0], fanout=4),
        StepSpec(WFK_WORKER, deps=[1]),
    ])
    pipe = mk_pipe([dag], replicate=NR, fail_ppt=250, drop_ppt=50,
                   max_retries=1)
    dlq = DLQStore()
    pipe.reset_runs()
    drained = 0
    for t in range(400):
        pipe.tick()
        pipe.readmit_succeeded()
        if t % 25 == 24:
            drained += pipe.drain_failed_to_dlq(dlq)
    drained += pipe.drain_failed_to_dlq(dlq)
    ok, fail = pipe.counts()
    active = pipe.active()
    admissions = NR + int(pipe.admit_count[0])
    assert ok > NR  # table cycled many times
    assert fail > 0  # max-retries exhaustion happened
    assert fail == drained  # every failure reached the host DLQ
    assert admissions == ok + fail + active  # nothing lost, nothing invented
    assert int(pipe.timeout_count[0]) > 0  # lost children were recovered
    assert int(pipe.retry_count[0]) > 0  # retry waves happened
    assert active == NR  # the table is still full (held concurrency)


def test_wf_property_random_dags_and_failures():
    """Property sweep: random DAGs × random fail/drop rates must always
    terminate (retry + timeout recovery guarantee progress) with exact
    accounting — no stuck steps, no lost children, every run terminal."""
    import random

    rng = random.Random(123)
    for trial in range(6):
        n_runs = rng.randint(2, 10)
        dags = []
        for _ in range(n_runs):
            n = rng.randint(1, 7)
            steps = []
            for s in range(n):
                deps = [d for d in range(s) if rng.random() < 0.4]
                kind = rng.choice([WFK_WORKER, WFK_FOR_EACH, WFK_CONDITION,
                                   WFK_DELAY])
                steps.append(StepSpec(kind, deps=deps,
                                      fanout=rng.randint(1, 12) if kind == WFK_FOR_EACH else 1,
                                      delay_ticks=rng.randint(0, 3),
                                      cond=rng.random() < 0.7))
            dags.append(DagSpec(steps=steps))
        pipe = mk_pipe(dags, fail_ppt=rng.choice([0, 100, 400]),
                       drop_ppt=rng.choice([0, 50, 150]),
                       max_retries=rng.randint(0, 3),
                       pad_cap=rng.choice([16, 64, 4096]))
        st = pipe.run_wave(max_ticks=600)
        ok, fail = pipe.counts()
        assert ok + fail == n_runs, (trial, ok, fail, n_runs)
        assert pipe.active() == 0, trial
        # no step left mid-flight on a terminal table
        assert int(pipe.children_out.abs().sum()) == 0 or int(pipe.rq_count[0]) >= 0
        out = pipe.children_out.view(-1, 64)
        states = pipe.step_state.view(-1, 64)
        for r in range(n_runs):
            for s in range(len(dags[r].steps)):
                stv = int(states[r, s])
                assert stv in (3, 4, 5, 0), (trial, r, s, stv)  # terminal or blocked-PENDING


def test_second_wave_redoes_real_work():
    """Regression pin for the CPU template-aliasing bug: .to(device) on CPU
    aliases the source tensor, so the live tables shared storage with the
    creation template and a re-admitted wave 'completed' instantly without
    emitting any children. Two waves on one pipeline must BOTH do the full
    child work."""
    FAN = 16
    pipe = mk_pipe([DagSpec(steps=[StepSpec(WFK_FOR_EACH, fanout=FAN)])] * 4)
    st1 = pipe.run_wave()
    assert st1.runs_succeeded == 4
    e1 = int(pipe.children_emitted.view(-1, 64)[:, 0].sum())
    assert e1 == 4 * FAN
    st2 = pipe.run_wave()
    assert st2.runs_succeeded == 4
    assert st2.ticks >= st1.ticks  # not an instant zombie wave
    e2 = int(pipe.children_emitted.view(-1, 64)[:, 0].sum())
    assert e2 == 4 * FAN  # the second wave re-emitted EVERY child
    done = pipe.children_done.view(-1, 64)
    assert all(int(done[r, 0]) == FAN for r in range(4))


def test_max_parallel_windows_inflight_children():
    """dataflow_test.go:71 semantics on device: a for_each with
    max_parallel=W never has more than W children in flight, the window
    slides as children land, and every child still runs exactly once."""
    FAN, W = 24, 4
    dag = DagSpec(steps=[StepSpec(WFK_FOR_EACH, fanout=FAN, max_parallel=W)])
    pipe = mk_pipe([dag] * 3, pad_cap=2)  # pad_cap 2: children land slowly
    pipe.reset_runs()
    max_seen = 0
    for _ in range(400):
        pipe.tick()
        out = pipe.children_out.view(-1, 64)[:, 0]
        max_seen = max(max_seen, int(out.max()))
        if pipe.active() == 0:
            break
    ok, fail = pipe.counts()
    assert ok == 3, (ok, fail)
    assert max_seen <= W          # the window held
    assert max_seen > 1           # and it actually pipelined
    done = pipe.children_done.view(-1, 64)
    assert all(int(done[r, 0]) == FAN for r in range(3))
