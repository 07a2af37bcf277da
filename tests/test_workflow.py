"""Workflow engine tests — ports of the reference behavioral oracle:
TestEngineForEachFanoutAndAggregateSuccess (workflow/engine_test.go:63),
TestEngineRetriesAndBackoff (:126), TestEngineApprovalPausesAndResumes (:191),
TestEngineDelayStepCompletes (:337), TestEngineConditionStepEvaluates (:430),
TestForEachMaxParallelLimitsDispatch (dataflow_test.go:71),
TestRerunFromCopiesDependencies / TestCancelRunPublishesCancels
(engine_extra_test.go:12,71), plus eval/template tests (eval_test.go)."""
import json

import pytest

from cordum_amd.bus import LoopbackBus
from cordum_amd.protocol import subjects as subj
from cordum_amd.protocol.capv2 import BusPacket, JobResult, JobStatus
from cordum_amd.store import MemoryStore
from cordum_amd.utils.clock import ManualClock
from cordum_amd.workflow import (
    Engine,
    RUN_FAILED,
    RUN_RUNNING,
    RUN_SUCCEEDED,
    RUN_WAITING,
    STEP_PENDING,
    STEP_RUNNING,
    STEP_SUCCEEDED,
    STEP_WAITING,
    Step,
    StepRun,
    Workflow,
    WorkflowRun,
    WorkflowStore,
    eval_expr,
    eval_template_string,
    eval_templates,
)
from cordum_amd.workflow.models import RUN_CANCELLED, STEP_FAILED


@pytest.fixture
def clock():
    return ManualClock()


def build(clock):
    bus = LoopbackBus(clock=clock)
    store = WorkflowStore(clock=clock)
    memory = MemoryStore(clock=clock)
    engine = Engine(store, bus, memory=memory, clock=clock)
    submitted = []
    bus.subscribe(subj.SUBJECT_SUBMIT, lambda s, p: submitted.append(p.job_request))
    cancels = []
    bus.subscribe(subj.SUBJECT_CANCEL, lambda s, p: cancels.append(p.job_cancel.job_id))
    events = []
    bus.subscribe(subj.SUBJECT_WORKFLOW_EVENT, lambda s, p: events.append(p.alert))
    return bus, store, memory, engine, submitted, cancels, events


def make_wf(store, steps, wf_id="wf1"):
    wf = Workflow(id=wf_id, org_id="org", steps={sid: Step.from_dict(sid, sd) for sid, sd in steps.items()})
    store.put_workflow(wf)
    return wf


def make_run(store, wf, input=None, run_id="run1"):
    run = WorkflowRun(id=run_id, workflow_id=wf.id, org_id="org", input=input or {})
    store.create_run(run)
    return run


def succeed(engine, memory, job_id, output=None):
    ptr = ""
    if output is not None:
        ptr = memory.put_result(job_id, json.dumps(output).encode())
    engine.handle_job_result(JobResult(job_id=job_id, status=JobStatus.SUCCEEDED, result_ptr=ptr))


def test_single_worker_step_lifecycle(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {"a": {"type": "worker", "topic": "job.echo"}})
    run = make_run(store, wf)
    engine.start_run(wf.id, run.id)
    assert len(submitted) == 1
    req = submitted[0]
    assert req.job_id == "run1:a@1"
    assert req.topic == "job.echo"
    assert req.env["run_id"] == "run1" and req.labels["step_id"] == "a"
    assert store.get_run("run1").status == RUN_RUNNING
    succeed(engine, memory, "run1:a@1", {"ok": True})
    run = store.get_run("run1")
    assert run.status == RUN_SUCCEEDED
    assert run.context["steps"]["a"]["output"] == {"ok": True}
    types = [e.type for e in store.get_timeline("run1")]
    assert "step_dispatched" in types and "step_completed" in types and "run_status" in types


def test_dependency_chain_waves(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "b": {"type": "worker", "topic": "job.t", "depends_on": ["a"]},
        "c": {"type": "worker", "topic": "job.t", "depends_on": ["b"]},
    })
    run = make_run(store, wf)
    engine.start_run(wf.id, run.id)
    assert [r.job_id for r in submitted] == ["run1:a@1"]
    succeed(engine, memory, "run1:a@1", {"v": 1})
    assert [r.job_id for r in submitted] == ["run1:a@1", "run1:b@1"]
    succeed(engine, memory, "run1:b@1", {"v": 2})
    succeed(engine, memory, "run1:c@1", {"v": 3})
    assert store.get_run("run1").status == RUN_SUCCEEDED


def test_failed_dep_blocks_downstream_and_fails_run(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "b": {"type": "worker", "topic": "job.t", "depends_on": ["a"]},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    engine.handle_job_result(JobResult(job_id="run1:a@1", status=JobStatus.FAILED, error_message="boom"))
    run = store.get_run("run1")
    assert run.status == RUN_FAILED
    assert run.steps["a"].status == STEP_FAILED
    assert len(submitted) == 1  # b never dispatched


def test_foreach_fanout_and_aggregate_success(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "fan": {"type": "worker", "topic": "job.t", "for_each": "${input.items}"},
    })
    make_run(store, wf, input={"items": ["x", "y", "z"]})
    engine.start_run(wf.id, "run1")
    assert len(submitted) == 3
    assert submitted[0].env["foreach_index"] == "0"
    assert json.loads(submitted[1].env["foreach_item"]) == "y"
    run = store.get_run("run1")
    assert run.steps["fan"].status == STEP_RUNNING
    assert set(run.steps["fan"].children) == {"fan[0]", "fan[1]", "fan[2]"}
    for i in range(3):
        succeed(engine, memory, f"run1:fan[{i}]@1", {"i": i})
    run = store.get_run("run1")
    assert run.steps["fan"].status == STEP_SUCCEEDED
    assert run.status == RUN_SUCCEEDED


def test_foreach_child_failure_fails_parent(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {"fan": {"type": "worker", "topic": "job.t", "for_each": "${input.items}"}})
    make_run(store, wf, input={"items": [1, 2]})
    engine.start_run(wf.id, "run1")
    succeed(engine, memory, "run1:fan[0]@1", {})
    engine.handle_job_result(JobResult(job_id="run1:fan[1]@1", status=JobStatus.FAILED, error_message="bad"))
    run = store.get_run("run1")
    assert run.steps["fan"].status == STEP_FAILED
    assert run.status == RUN_FAILED


def test_foreach_max_parallel_limits_dispatch(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "fan": {"type": "worker", "topic": "job.t", "for_each": "${input.items}", "max_parallel": 2},
    })
    make_run(store, wf, input={"items": list(range(5))})
    engine.start_run(wf.id, "run1")
    assert len(submitted) == 2  # window
    run = store.get_run("run1")
    assert sum(1 for c in run.steps["fan"].children.values() if c.status == STEP_RUNNING) == 2
    assert sum(1 for c in run.steps["fan"].children.values() if c.status == STEP_PENDING) == 3
    succeed(engine, memory, "run1:fan[0]@1", {})
    assert len(submitted) == 3  # next child dispatched as capacity frees
    succeed(engine, memory, "run1:fan[1]@1", {})
    succeed(engine, memory, "run1:fan[2]@1", {})
    succeed(engine, memory, "run1:fan[3]@1", {})
    succeed(engine, memory, "run1:fan[4]@1", {})
    assert store.get_run("run1").status == RUN_SUCCEEDED
    assert len(submitted) == 5


def test_retries_and_backoff(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t",
              "retry": {"max_retries": 2, "initial_backoff_sec": 2, "multiplier": 2}},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    engine.handle_job_result(JobResult(job_id="run1:a@1", status=JobStatus.FAILED, error_message="e1"))
    run = store.get_run("run1")
    assert run.steps["a"].status == STEP_PENDING
    assert run.steps["a"].next_attempt_at == pytest.approx(clock.now() + 2)
    assert len(submitted) == 1  # backoff window holds
    clock.advance(2.1)
    assert engine.pump_timers() == 1
    assert len(submitted) == 2
    assert submitted[1].job_id == "run1:a@2"
    # second failure: backoff 2 * 2^(2-1) = 4s
    engine.handle_job_result(JobResult(job_id="run1:a@2", status=JobStatus.FAILED, error_message="e2"))
    run = store.get_run("run1")
    assert run.steps["a"].next_attempt_at == pytest.approx(clock.now() + 4)
    clock.advance(4.1)
    engine.pump_timers()
    assert len(submitted) == 3
    # third failure: attempts=3 > max_retries=2 -> terminal failure
    engine.handle_job_result(JobResult(job_id="run1:a@3", status=JobStatus.FAILED, error_message="e3"))
    run = store.get_run("run1")
    assert run.steps["a"].status == STEP_FAILED
    assert run.status == RUN_FAILED


def test_approval_pauses_and_resumes(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "gate": {"type": "approval"},
        "after": {"type": "worker", "topic": "job.t", "depends_on": ["gate"]},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    run = store.get_run("run1")
    assert run.status == RUN_WAITING
    assert run.steps["gate"].status == STEP_WAITING
    assert len(submitted) == 0
    engine.approve_step("run1", "gate", approved=True)
    assert len(submitted) == 1
    succeed(engine, memory, "run1:after@1", {})
    assert store.get_run("run1").status == RUN_SUCCEEDED


def test_approval_rejection_fails_run(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {"gate": {"type": "approval"}})
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    engine.approve_step("run1", "gate", approved=False)
    assert store.get_run("run1").status == RUN_FAILED


def test_delay_step_completes(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "wait": {"type": "delay", "delay_sec": 10},
        "after": {"type": "worker", "topic": "job.t", "depends_on": ["wait"]},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    run = store.get_run("run1")
    assert run.steps["wait"].status == STEP_RUNNING
    assert len(submitted) == 0
    clock.advance(10.5)
    assert engine.pump_timers() == 1
    run = store.get_run("run1")
    assert run.steps["wait"].status == STEP_SUCCEEDED
    assert len(submitted) == 1


def test_condition_step_evaluates(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "check": {"type": "condition", "condition": "input.n > 3"},
        "yes": {"type": "worker", "topic": "job.t", "depends_on": ["check"],
                "condition": "steps.check.output == true"},
    })
    make_run(store, wf, input={"n": 5})
    engine.start_run(wf.id, "run1")
    run = store.get_run("run1")
    assert run.steps["check"].status == STEP_SUCCEEDED
    assert run.steps["check"].output is True
    assert run.context["steps"]["check"]["output"] is True
    assert len(submitted) == 1  # 'yes' dispatched


def test_condition_gate_skips_step(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "maybe": {"type": "worker", "topic": "job.t", "condition": "input.go == true"},
    })
    make_run(store, wf, input={"go": False})
    engine.start_run(wf.id, "run1")
    run = store.get_run("run1")
    assert run.steps["maybe"].status == STEP_SUCCEEDED  # auto-succeed on skip
    assert len(submitted) == 0
    assert run.status == RUN_SUCCEEDED


def test_notify_step_emits_alert(clock):
    bus, store, memory, engine, submitted, cancels, events = build(clock)
    wf = make_wf(store, {
        "n": {"type": "notify", "input": {"severity": "warn", "message": "hello ${input.name}"}},
    })
    make_run(store, wf, input={"name": "world"})
    engine.start_run(wf.id, "run1")
    assert len(events) == 1
    assert events[0].severity == "warn"
    assert events[0].message == "hello world"
    assert store.get_run("run1").status == RUN_SUCCEEDED


def test_rerun_from_copies_dependencies(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "b": {"type": "worker", "topic": "job.t", "depends_on": ["a"]},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    succeed(engine, memory, "run1:a@1", {"from": "a"})
    engine.handle_job_result(JobResult(job_id="run1:b@1", status=JobStatus.FAILED, error_message="x"))
    assert store.get_run("run1").status == RUN_FAILED
    new_id = engine.rerun_from("run1", "b")
    new_run = store.get_run(new_id)
    assert new_run.rerun_of == "run1" and new_run.rerun_step == "b"
    assert new_run.steps["a"].status == STEP_SUCCEEDED  # dep cloned
    assert new_run.context["steps"]["a"]["output"] == {"from": "a"}
    engine.start_run(wf.id, new_id)
    assert submitted[-1].job_id == f"{new_id}:b@1"  # only b dispatched


def test_rerun_requires_succeeded_dep(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "b": {"type": "worker", "topic": "job.t", "depends_on": ["a"]},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    engine.handle_job_result(JobResult(job_id="run1:a@1", status=JobStatus.FAILED))
    with pytest.raises(ValueError):
        engine.rerun_from("run1", "b")


def test_cancel_run_publishes_cancels(clock):
    bus, store, memory, engine, submitted, cancels, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "fan": {"type": "worker", "topic": "job.t", "for_each": "${input.items}"},
    })
    make_run(store, wf, input={"items": [1, 2]})
    engine.start_run(wf.id, "run1")
    assert len(submitted) == 3
    engine.cancel_run("run1")
    run = store.get_run("run1")
    assert run.status == RUN_CANCELLED
    assert set(cancels) == {"run1:a@1", "run1:fan[0]@1", "run1:fan[1]@1"}
    # results for cancelled run ignored
    succeed(engine, memory, "run1:a@1", {})
    assert store.get_run("run1").status == RUN_CANCELLED


def test_duplicate_result_ignored(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "b": {"type": "worker", "topic": "job.t", "depends_on": ["a"]},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    succeed(engine, memory, "run1:a@1", {"v": 1})
    n = len(submitted)
    succeed(engine, memory, "run1:a@1", {"v": 99})  # duplicate
    assert len(submitted) == n
    assert store.get_run("run1").context["steps"]["a"]["output"] == {"v": 1}


def test_output_schema_validation_fails_step(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t",
              "output_schema": {"type": "object", "required": ["ok"]}},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    succeed(engine, memory, "run1:a@1", {"wrong": 1})
    run = store.get_run("run1")
    assert run.steps["a"].status == STEP_FAILED
    assert run.status == RUN_FAILED
    assert any(e.type == "step_output_invalid" for e in store.get_timeline("run1"))


def test_output_path_applied(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t", "output_path": "results.first"},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    succeed(engine, memory, "run1:a@1", {"x": 1})
    run = store.get_run("run1")
    assert run.context["results"]["first"] == {"x": 1}


def test_step_input_templates(clock):
    bus, store, memory, engine, submitted, _, _ = build(clock)
    wf = make_wf(store, {
        "a": {"type": "worker", "topic": "job.t"},
        "b": {"type": "worker", "topic": "job.t", "depends_on": ["a"],
              "input": {"prev": "${steps.a.output.v}", "greeting": "v=${steps.a.output.v}!", "n": 42}},
    })
    make_run(store, wf)
    engine.start_run(wf.id, "run1")
    succeed(engine, memory, "run1:a@1", {"v": 7})
    req = submitted[-1]
    payload = json.loads(memory.get_pointer(req.context_ptr))
    assert payload == {"prev": 7, "greeting": "v=7!", "n": 42}


# --- eval / templates ---------------------------------------------------------


@pytest.mark.parametrize(
    "expr,scope,want",
    [
        ("input.n > 3", {"input": {"n": 5}}, True),
        ("input.n >= 5", {"input": {"n": 5}}, True),
        ("input.s == 'abc'", {"input": {"s": "abc"}}, True),
        ("!input.flag", {"input": {"flag": False}}, True),
        ("length(input.items) == 2", {"input": {"items": [1, 2]}}, True),
        ("first(input.items)", {"input": {"items": ["a", "b"]}}, "a"),
        ("length(missing.path)", {}, 0),
        ("true", {}, True),
        ("3.5", {}, 3.5),
        ("input.x != 1", {"input": {"x": 2}}, True),
    ],
)
def test_eval_matrix(expr, scope, want):
    assert eval_expr(expr, scope) == want


def test_template_type_preservation():
    scope = {"input": {"n": 3, "items": [1, 2]}}
    assert eval_template_string("${input.n}", scope) == 3.0 or eval_template_string("${input.n}", scope) == 3
    assert eval_template_string("n is ${input.n}", scope) == "n is 3"
    assert eval_template_string("no templates", scope) == "no templates"
    assert eval_templates({"a": "${input.items}", "b": ["${input.n}"]}, scope) == {"a": [1, 2], "b": [3]}
    with pytest.raises(Exception):
        eval_template_string("${unterminated", scope)
