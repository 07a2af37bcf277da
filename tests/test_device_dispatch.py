"""Device dispatch engine (runtime/device_dispatch.py): the batched K1/K2
serving path must be semantically identical to the host scheduler engine.

All tests run the CPU reference backend (ops/pipeline._RefOps) — the same
flush/route code the GPU runs with the HIP extension; kernel==reference
bit-equality is covered by tests/test_gpu_kernels.py on the GPU box.
"""
import random

import pytest

from cordum_amd.protocol import JobState
from cordum_amd.protocol.capv2 import Heartbeat, JobMetadata, JobRequest
from cordum_amd.runtime.node import Node
from cordum_amd.scheduler import PoolProfile, PoolRouting
from cordum_amd.utils.clock import ManualClock

POLICY = """
version: v-test
rules:
  - id: deny-risky
    match: {risk_tags: [dangerous]}
    decision: deny
    reason: dangerous work
  - id: approve-prod
    match: {topics: ["job.prod.*"]}
    decision: require_approval
    reason: prod gate
  - id: throttle-bulk
    match: {tenants: [bulk]}
    decision: throttle
    reason: bulk tenant
  - id: constrain-team
    match: {labels: {team: sec}}
    decision: allow_with_constraints
    constraints:
      budgets: {max_runtime_ms: 5000}
  - id: allow-echo
    match: {topics: ["job.echo"]}
    decision: allow
"""

ROUTING = PoolRouting(
    topics={
        "job.default": ["default"],
        "job.echo": ["default"],
        "job.prod.deploy": ["default"],
        "job.gpu": ["gpu"],
    },
    pools={"default": PoolProfile(), "gpu": PoolProfile(requires=["mfma"])},
)


def make_node(dispatch, policy=POLICY, workers=1):
    clock = ManualClock()
    n = Node(clock=clock, routing=ROUTING, policy_yaml=policy,
             dispatch=dispatch, backend="ref" if dispatch == "device" else None).start()
    for i in range(workers):
        n.add_worker(f"w{i}", topics=["job.default", "job.echo", "job.prod.deploy"])
    return n


def submit(n, job_id, topic="job.echo", **kw):
    ctx = kw.pop("context", b"{}")
    req = JobRequest(job_id=job_id, topic=topic, tenant_id=kw.pop("tenant", "default"), **kw)
    n.submit_job(req, context=ctx)
    n.drain()
    return n.job_store.get_state(job_id)


# ---------------------------------------------------------------- decision parity
def _random_req(rng, i):
    topic = rng.choice(["job.echo", "job.default", "job.prod.deploy", "job.other"])
    labels = {}
    if rng.random() < 0.3:
        labels["team"] = rng.choice(["sec", "eng"])
    meta = JobMetadata(risk_tags=["dangerous"] if rng.random() < 0.3 else [])
    return JobRequest(job_id=f"j{i}", topic=topic, tenant_id=rng.choice(["default", "bulk"]),
                      labels=labels, meta=meta)


def test_decision_and_state_parity_host_vs_device():
    """200 random jobs through both engines: final states and persisted
    safety-decision records must be identical."""
    rng = random.Random(42)
    reqs = [_random_req(rng, i) for i in range(200)]
    nodes = {d: make_node(d) for d in ("host", "device")}
    states, records = {}, {}
    for d, n in nodes.items():
        for req in reqs:
            n.submit_job(JobRequest.decode(req.encode()), context=b"{}")
        n.drain()
        states[d] = [n.job_store.get_state(r.job_id) for r in reqs]
        records[d] = [n.job_store.get_safety_decision(r.job_id) for r in reqs]
    assert states["host"] == states["device"]
    for rh, rd in zip(records["host"], records["device"]):
        if rh is None:
            assert rd is None
            continue
        assert (rh.decision, rh.rule_id, rh.reason, rh.policy_snapshot, rh.job_hash) == (
            rd.decision, rd.rule_id, rd.reason, rd.policy_snapshot, rd.job_hash)
    # the device node actually used the batched kernel path
    assert nodes["device"].device_gate.jobs_evaluated >= 100
    assert nodes["device"].scheduler.device_routed > 0


def test_deny_goes_to_dlq_device():
    n = make_node("device")
    meta = JobMetadata(risk_tags=["dangerous"])
    assert submit(n, "d1", meta=meta) == JobState.DENIED
    entries, _ = n.dlq.list()
    assert any(e.job_id == "d1" and e.reason_code == "safety_denied" for e in entries)


def test_approval_flow_device():
    """require_approval -> APPROVAL_REQUIRED with bound hash; approval
    label honored on resubmit only when the hash matches (engine.go:484-522)."""
    n = make_node("device")
    req = JobRequest(job_id="p1", topic="job.prod.deploy", tenant_id="default")
    n.submit_job(req, context=b"{}")
    n.drain()
    assert n.job_store.get_state("p1") == JobState.APPROVAL_REQUIRED
    rec = n.job_store.get_safety_decision("p1")
    assert rec.decision == "require_approval" and rec.job_hash

    resub = n.job_store.get_job_request("p1")
    resub.labels = dict(resub.labels)
    resub.labels["approval_granted"] = "true"
    resub.labels["cordum.bus_msg_id"] = "resub-1"
    n.submit_job(resub)
    n.drain()
    assert n.job_store.get_state("p1") == JobState.SUCCEEDED

    # tampered resubmit: hash mismatch -> approval ignored -> approval again
    req2 = JobRequest(job_id="p2", topic="job.prod.deploy", tenant_id="default")
    n.submit_job(req2, context=b"{}")
    n.drain()
    tam = n.job_store.get_job_request("p2")
    tam.labels = dict(tam.labels)
    tam.labels["approval_granted"] = "true"
    tam.labels["cordum.bus_msg_id"] = "resub-2"
    tam.env = dict(tam.env or {})
    tam.env["INJECTED"] = "x"  # changes the job hash
    n.submit_job(tam)
    n.drain()
    assert n.job_store.get_state("p2") == JobState.APPROVAL_REQUIRED


def test_throttle_requeues_with_delay_device():
    n = make_node("device")
    req = JobRequest(job_id="t1", topic="job.echo", tenant_id="bulk")
    n.submit_job(req, context=b"{}")
    n.drain()
    # throttled: not dispatched, requeued on the delayed heap
    assert n.job_store.get_state("t1") == JobState.PENDING
    assert n.bus.pending_count() > 0
    # advance past the throttle delay: redelivered, throttled again (policy
    # unchanged), until the delivery budget is spent -> stays PENDING
    for _ in range(10):
        n.clock.advance(6.0)
        n.drain()
    assert n.job_store.get_state("t1") == JobState.PENDING
    assert n.bus.pending_count() == 0  # budget exhausted, no hot loop


def test_no_workers_retry_then_dispatch_device():
    n = make_node("device", workers=0)
    req = JobRequest(job_id="n1", topic="job.echo", tenant_id="default")
    n.submit_job(req, context=b'{"a": 1}')
    n.drain()
    assert n.job_store.get_state("n1") == JobState.PENDING
    n.add_worker("late", topics=["job.echo"])
    n.clock.advance(3.0)
    n.drain()
    assert n.job_store.get_state("n1") == JobState.SUCCEEDED


def test_requires_filtering_no_pool_fails_device():
    n = make_node("device")
    meta = JobMetadata(requires=["quantum"])  # no pool satisfies
    assert submit(n, "r1", topic="job.gpu", meta=meta) == JobState.FAILED
    assert any(e.job_id == "r1" and e.reason_code == "no_pool_mapping" for e in n.dlq.list()[0])


def test_placement_label_routing_device():
    """Job with a placement label must land on the matching worker only —
    exercised through the K2 label-mask path."""
    n = make_node("device", workers=0)
    n.add_worker("plain", topics=["job.echo"])
    n.add_worker("labeled", topics=["job.echo"], labels={"zone": "eu"})
    req = JobRequest(job_id="L1", topic="job.echo", tenant_id="default",
                     labels={"zone": "eu"})
    n.submit_job(req, context=b"{}")
    n.drain()
    assert n.job_store.get_state("L1") == JobState.SUCCEEDED
    assert n.job_store.get_job_meta("L1").get("worker_id") == "labeled"

    # label no worker carries -> IMPOSSIBLE bit -> no_workers retry, PENDING
    req2 = JobRequest(job_id="L2", topic="job.echo", tenant_id="default",
                      labels={"zone": "mars"})
    n.submit_job(req2, context=b"{}")
    n.drain()
    assert n.job_store.get_state("L2") == JobState.PENDING


def test_routing_pick_parity_random_fleets():
    """DeviceWorkerTable.pick == LeastLoadedStrategy.pick_subject on random
    fleets (the K2 oracle is the strategy's scoring loop)."""
    import torch  # noqa: F401

    from cordum_amd.ops.pipeline import _RefOps
    from cordum_amd.ops.worker_table import DeviceWorkerTable
    from cordum_amd.scheduler.strategy import LeastLoadedStrategy

    rng = random.Random(7)
    strat = LeastLoadedStrategy(ROUTING)
    for trial in range(20):
        workers = {}
        for i in range(rng.randint(1, 30)):
            wid = f"w{i:02d}"
            workers[wid] = Heartbeat(
                worker_id=wid,
                pool=rng.choice(["default", "gpu"]),
                active_jobs=rng.randint(0, 10),
                max_parallel_jobs=rng.choice([0, 4, 10]),
                cpu_load=rng.random() * 100,
                gpu_utilization=rng.random() * 100,
                labels={"zone": rng.choice(["eu", "us"])} if rng.random() < 0.4 else {},
            )
        table = DeviceWorkerTable("cpu", _RefOps())
        assert table.pack(workers)
        for topic in ("job.echo", "job.default"):
            req = JobRequest(job_id="x", topic=topic, tenant_id="default",
                             labels={"zone": "eu"} if rng.random() < 0.3 else {})
            eligible, req_labels, _ = strat.resolve(req)
            pick = table.pick([table.pool_mask(eligible)], [table.label_mask(req_labels)])
            idx = int(pick[0])
            try:
                subject = strat.pick_subject(req, workers)
                assert idx >= 0, f"host picked {subject}, device returned {idx}"
                assert subject == f"worker.{table.worker_ids[idx]}.jobs"
            except Exception as e:
                code = getattr(e, "reason_code", "")
                assert idx < 0
                if code == "pool_overloaded":
                    assert idx == -2
                else:
                    assert idx == -1


def test_device_pool_cancel_and_oversize():
    n = make_node("device", workers=0)
    pool = n.add_device_worker_pool(n_workers=2, topics=["job.echo"],
                                    max_payload_bytes=64)
    # oversize payload fails loudly, not silently
    big = b"x" * 128
    req = JobRequest(job_id="big", topic="job.echo", tenant_id="default")
    n.submit_job(req, context=big)
    n.drain()
    assert n.job_store.get_state("big") == JobState.FAILED
    assert any(e.job_id == "big" for e in n.dlq.list()[0])

    # normal payload round-trips through the pool arena
    req2 = JobRequest(job_id="ok", topic="job.echo", tenant_id="default")
    n.submit_job(req2, context=b"hello world")
    n.drain()
    assert n.job_store.get_state("ok") == JobState.SUCCEEDED
    ptr = n.job_store.get_job_meta("ok")["result_ptr"]
    assert n.memory.get_pointer(ptr) == b"hello world"
    assert pool.jobs_executed == 1


def test_policy_hot_reload_recompiles_gate():
    """Publishing a policy bundle through the config service must swap the
    device gate's compiled tensors (kernel watch -> on_policy_swap)."""
    n = make_node("device")
    # job.default matches no base rule -> default allow
    assert submit(n, "h1", topic="job.default") == JobState.SUCCEEDED
    bundle = "rules:\n  - id: block-default\n    match: {topics: ['job.default']}\n    decision: deny\n    reason: blocked\n"
    doc = {"enabled": True, "bundles": {"b1": {"enabled": True, "content": bundle}}}
    n.configsvc.set("system", "policy", doc)
    assert submit(n, "h2", topic="job.default") == JobState.DENIED
    rec = n.job_store.get_safety_decision("h2")
    assert rec.rule_id == "block-default"
    assert n.device_gate.device_active
    # base rules still win first-match over appended fragments
    assert submit(n, "h3", topic="job.echo") == JobState.SUCCEEDED


def test_gateway_submit_through_device_engine():
    """POST /api/v1/jobs -> device engine -> device pool -> result readable
    through the API (the round-1 'serving path never touches the GPU' gap,
    CPU reference backend here; same path runs the HIP kernels on GPU)."""
    fastapi = pytest.importorskip("fastapi")  # noqa: F841
    from fastapi.testclient import TestClient

    from cordum_amd.gateway.app import create_app

    n = make_node("device", workers=0)
    n.add_device_worker_pool(n_workers=2, topics=["job.echo", "job.default"])
    app = create_app(n)
    c = TestClient(app)
    c.headers.update({"X-API-Key": "test-key", "X-Principal-Role": "admin"})
    r = c.post("/api/v1/jobs", json={"prompt": "hello gpu", "topic": "job.echo"})
    assert r.status_code == 200, r.text
    job_id = r.json()["job_id"]
    d = c.get(f"/api/v1/jobs/{job_id}").json()
    assert d["state"] == "SUCCEEDED"
    assert "hello gpu" in str(d.get("result", ""))
    assert n.device_gate.jobs_evaluated >= 1
    assert n.scheduler.device_routed >= 1


def test_cancel_before_flush_drops_buffered_job():
    """A job cancelled while buffered in the device engine must not be
    dispatched — and must not abort the rest of the flush batch (this
    crashed the whole drain with InvalidTransition before the fix)."""
    from cordum_amd.protocol import JobState
    from cordum_amd.protocol.capv2 import JobRequest
    from cordum_amd.runtime.node import Node
    from cordum_amd.utils.clock import ManualClock

    node = Node(clock=ManualClock(), dispatch="device", backend="ref").start()
    node.add_device_worker_pool(n_workers=2, topics=["job.default"])
    node.submit_job(JobRequest(job_id="cx-1", topic="job.default",
                               tenant_id="default"), context=b"{}")
    node.submit_job(JobRequest(job_id="cx-2", topic="job.default",
                               tenant_id="default"), context=b"{}")
    assert node.scheduler.pending_count() == 2
    assert node.scheduler.cancel_job("cx-1")
    node.drain()
    assert node.job_store.get_state("cx-1") == JobState.CANCELLED
    assert node.job_store.get_state("cx-2") == JobState.SUCCEEDED


def test_pending_replayer_redrives_device_engine():
    """A job that exhausts its redelivery budget (no workers) parks
    PENDING; when capacity appears, the pending replayer re-drives it
    through the batched device path (pending_replayer.go semantics)."""
    from cordum_amd.protocol import JobState
    from cordum_amd.protocol.capv2 import JobRequest
    from cordum_amd.runtime.node import Node
    from cordum_amd.scheduler import PoolProfile, PoolRouting
    from cordum_amd.utils.clock import ManualClock

    clock = ManualClock()
    routing = PoolRouting(topics={"job.default": ["default"]},
                          pools={"default": PoolProfile()})
    node = Node(clock=clock, routing=routing, dispatch="device",
                backend="ref").start()
    node.submit_job(JobRequest(job_id="p1", topic="job.default",
                               tenant_id="default"), context=b"{}")
    node.drain()
    for _ in range(10):  # exhaust the NAK budget against an empty fleet
        clock.advance(5)
        node.tick()
    assert node.job_store.get_state("p1") == JobState.PENDING
    assert not node.scheduler._redeliveries  # budget spent, not looping

    node.add_device_worker_pool(n_workers=2, topics=["job.default"])
    clock.advance(400)  # past the replayer age threshold
    node.reconcile()
    node.drain()
    for _ in range(6):
        clock.advance(5)
        node.tick()
    assert node.job_store.get_state("p1") == JobState.SUCCEEDED
