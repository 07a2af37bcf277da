"""End-to-end single-process node tests — BASELINE config #1:
the hello-pack echo workflow on CPU (in-process bus, in-memory state,
one worker). Oracle flow: SURVEY.md §3.1/§3.2; hello-pack:
examples/hello-pack/pack.yaml (echo workflow + schema)."""
import json

import pytest

from cordum_amd.protocol import JobState
from cordum_amd.protocol.capv2 import JobRequest
from cordum_amd.runtime.node import Node
from cordum_amd.scheduler import PoolProfile, PoolRouting
from cordum_amd.utils.clock import ManualClock
from cordum_amd.workflow import RUN_SUCCEEDED, RUN_WAITING, Step, Workflow, WorkflowRun


@pytest.fixture(params=["host", "device"])
def node(request):
    """Every e2e flow runs twice: host dispatch engine and the device
    dispatch engine (K1/K2 batched path, CPU reference backend in CI —
    the identical code path the GPU runs with the HIP extension)."""
    clock = ManualClock()
    routing = PoolRouting(
        topics={"job.default": ["default"], "job.echo": ["default"]},
        pools={"default": PoolProfile()},
    )
    n = Node(clock=clock, routing=routing, dispatch=request.param,
             backend="ref" if request.param == "device" else None).start()
    n.add_worker("w1", topics=["job.default", "job.echo"])
    return n


def test_echo_job_end_to_end(node):
    req = JobRequest(job_id="job-1", topic="job.echo", tenant_id="default")
    node.submit_job(req, context=b'{"message": "hello"}')
    node.drain()
    assert node.job_store.get_state("job-1") == JobState.SUCCEEDED
    ptr = node.job_store.get_job_meta("job-1")["result_ptr"]
    assert json.loads(node.memory.get_pointer(ptr)) == {"message": "hello"}


def test_hello_pack_echo_workflow(node):
    """hello-pack: input {message} -> echo step -> run succeeds with output."""
    wf = Workflow(
        id="hello-workflow",
        org_id="default",
        name="Hello",
        input_schema={"type": "object", "required": ["message"], "properties": {"message": {"type": "string"}}},
        steps={
            "echo": Step.from_dict("echo", {
                "type": "worker",
                "topic": "job.echo",
                "input": {"message": "${input.message}"},
            }),
        },
    )
    node.workflow_store.put_workflow(wf)
    run = WorkflowRun(id="run-1", workflow_id=wf.id, org_id="default", input={"message": "hi there"})
    node.workflow_store.create_run(run)
    node.workflow.start_run(wf.id, "run-1")
    node.drain()
    run = node.workflow_store.get_run("run-1")
    assert run.status == RUN_SUCCEEDED
    assert run.context["steps"]["echo"]["output"] == {"message": "hi there"}
    # job side: the step job went through the full scheduler state machine
    assert node.job_store.get_state("run-1:echo@1") == JobState.SUCCEEDED


def test_workflow_approval_gate_with_worker_steps(node):
    wf = Workflow(
        id="gated",
        org_id="default",
        steps={
            "prep": Step.from_dict("prep", {"type": "worker", "topic": "job.default"}),
            "gate": Step.from_dict("gate", {"type": "approval", "depends_on": ["prep"]}),
            "ship": Step.from_dict("ship", {"type": "worker", "topic": "job.default", "depends_on": ["gate"]}),
        },
    )
    node.workflow_store.put_workflow(wf)
    node.workflow_store.create_run(WorkflowRun(id="r2", workflow_id="gated", org_id="default", input={"x": 1}))
    node.workflow.start_run("gated", "r2")
    node.drain()
    assert node.workflow_store.get_run("r2").status == RUN_WAITING
    node.workflow.approve_step("r2", "gate", approved=True)
    node.drain()
    assert node.workflow_store.get_run("r2").status == RUN_SUCCEEDED


def test_fanout_workflow_through_real_scheduler(node):
    wf = Workflow(
        id="fan",
        org_id="default",
        steps={"fan": Step.from_dict("fan", {"type": "worker", "topic": "job.default",
                                             "for_each": "${input.items}"})},
    )
    node.workflow_store.put_workflow(wf)
    items = [{"i": i} for i in range(16)]
    node.workflow_store.create_run(WorkflowRun(id="r3", workflow_id="fan", org_id="default", input={"items": items}))
    node.workflow.start_run("fan", "r3")
    node.drain()
    run = node.workflow_store.get_run("r3")
    assert run.status == RUN_SUCCEEDED
    assert len([s for s in run.steps if s.startswith("fan[")]) == 16


def test_dlq_tap_and_retry_path(node):
    # no routing for this topic -> dispatch fails -> DLQ entry written by tap
    req = JobRequest(job_id="bad-1", topic="job.unrouted", tenant_id="default")
    node.submit_job(req, context=b"{}")
    node.drain()
    assert node.job_store.get_state("bad-1") == JobState.FAILED
    entry = node.dlq.get("bad-1")
    assert entry is not None and entry.reason_code == "no_pool_mapping"


def test_routing_hot_swap_via_config_overlay(node):
    req = JobRequest(job_id="hs-1", topic="job.newpool", tenant_id="default")
    node.submit_job(req, context=b"{}")
    node.drain()
    assert node.job_store.get_state("hs-1") == JobState.FAILED  # unrouted
    # install routing through the config overlay (pack-style)
    node.configsvc.set("system", "default", {
        "pools": {"topics": {"job.newpool": ["default"], "job.default": ["default"], "job.echo": ["default"]},
                  "pools": {"default": {}}},
    })
    req2 = JobRequest(job_id="hs-2", topic="job.newpool", tenant_id="default")
    node.submit_job(req2, context=b"{}")
    node.drain()
    assert node.job_store.get_state("hs-2") == JobState.SUCCEEDED


def test_policy_node_approval_flow():
    clock = ManualClock()
    routing = PoolRouting(topics={"job.default": ["default"]}, pools={"default": PoolProfile()})
    policy = """
version: v1
default_tenant: default
rules:
  - id: approve-deploys
    decision: require_approval
    reason: deploys need a human
    match: {topics: ["job.default"], risk_tags: [deploy]}
"""
    n = Node(clock=clock, routing=routing, policy_yaml=policy).start()
    n.add_worker("w1")
    from cordum_amd.protocol.capv2 import JobMetadata
    req = JobRequest(job_id="dep-1", topic="job.default", tenant_id="default",
                     meta=JobMetadata(risk_tags=["deploy"]))
    n.submit_job(req, context=b"{}")
    n.drain()
    assert n.job_store.get_state("dep-1") == JobState.APPROVAL_REQUIRED
    dec = n.job_store.get_safety_decision("dep-1")
    assert dec.decision == "require_approval" and dec.job_hash

    # approve via the gateway path semantics: label + fresh msg id + PENDING + republish
    stored = n.job_store.get_job_request("dep-1")
    stored.labels["approval_granted"] = "true"
    stored.labels["cordum.bus_msg_id"] = "approval:1"
    n.job_store.set_state("dep-1", JobState.PENDING)
    n.submit_job(stored, context=None)
    n.drain()
    assert n.job_store.get_state("dep-1") == JobState.SUCCEEDED


def test_worker_crash_timeout_dlq_and_retry():
    """Failure detection end-to-end (reconciler.go:88-144 + dlq retry
    gateway.go:3452): a worker receives a job and never reports back; the
    staleness reconciler times it out, the DLQ records it, and the retry
    path re-runs it to success once a healthy worker serves the topic."""
    from cordum_amd.protocol.capv2 import BusPacket, Heartbeat
    from cordum_amd.protocol import subjects as subj

    clock = ManualClock()
    routing = PoolRouting(topics={"job.default": ["default"]},
                          pools={"default": PoolProfile()})
    n = Node(clock=clock, routing=routing).start()

    # a "worker" that heartbeats (so routing picks it) but swallows jobs
    swallowed = []
    n.bus.subscribe(subj.worker_subject("zombie"),
                    lambda s, p: swallowed.append(p.job_request.job_id))
    n.bus.publish(subj.SUBJECT_HEARTBEAT, BusPacket(
        protocol_version=1,
        heartbeat=Heartbeat(worker_id="zombie", pool="default",
                            max_parallel_jobs=4)))
    n.drain()

    n.submit_job(JobRequest(job_id="lost-1", topic="job.default",
                            tenant_id="default"), context=b'{"p": 1}')
    n.drain()
    assert swallowed == ["lost-1"]
    from cordum_amd.protocol import JobState as JS

    assert n.job_store.get_state("lost-1") in (JS.DISPATCHED, JS.RUNNING)

    # past the dispatch/running cutoffs the reconciler declares TIMEOUT
    # (the reference reconciler only transitions state — no DLQ entry,
    # reconciler.go:88-129 — clients observe TIMEOUT and resubmit/remediate)
    clock.advance(10_000)
    n.reconcile()
    n.drain()
    assert n.job_store.get_state("lost-1") == JS.TIMEOUT
    events = n.job_store.get_events("lost-1")
    assert any(e.endswith("|TIMEOUT") for e in events)

    # the zombie's heartbeat has expired with the clock advance; a healthy
    # worker joins and the resubmission runs to completion
    n.add_worker("healthy", topics=["job.default"])
    n.submit_job(JobRequest(job_id="lost-1-resubmit", topic="job.default",
                            tenant_id="default"), context=b'{"p": 1}')
    n.drain()
    assert n.job_store.get_state("lost-1-resubmit") == JS.SUCCEEDED
    assert swallowed == ["lost-1"]  # the zombie never got the resubmission


def test_remediation_reroutes_and_dispatches(node):
    """Remediation must produce a job that actually ROUTES: the audit
    labels the gateway stamps (remediation_of/remediation_id) are not
    placement constraints (reference bug: strategy_least_loaded.go's
    filter keeps them, so remediated jobs are unroutable there; documented
    deviation like the approval_* family)."""
    from cordum_amd.protocol.capv2 import PolicyRemediation
    from cordum_amd.store.job_store import SafetyDecisionRecord

    req = JobRequest(job_id="rm-1", topic="job.echo", tenant_id="default",
                     labels={"remediation_of": "old-1", "remediation_id": "r1",
                             "retry": "true", "dlq_entry": "old-1",
                             "retry_of_job": "old-1"})
    node.submit_job(req, context=b'{"m": 1}')
    node.drain()
    assert node.job_store.get_state("rm-1") == JobState.SUCCEEDED
