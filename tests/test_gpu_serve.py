"""GPU serving-path tests: API-submitted jobs decided/routed/executed by the
HIP kernels (K1 policy gate, K2 least-loaded, echo pool) on a real MI355X.

These are the same flows tests/test_device_dispatch.py runs on the CPU
reference backend; here the Node uses backend="ext" so every batched call
lands in cordum_hip_ops (and fails loudly if the extension is missing —
ops/__init__.get_ext(required=True))."""
import pytest

from cordum_amd.protocol import JobState
from cordum_amd.protocol.capv2 import JobMetadata, JobRequest
from cordum_amd.runtime.node import Node
from cordum_amd.scheduler import PoolProfile, PoolRouting
from cordum_amd.utils.clock import ManualClock

pytestmark = pytest.mark.gpu

POLICY = """
version: v-gpu
rules:
  - id: deny-risky
    match: {risk_tags: [dangerous]}
    decision: deny
    reason: dangerous work
  - id: approve-prod
    match: {topics: ["job.prod.*"]}
    decision: require_approval
    reason: prod gate
  - id: allow-echo
    match: {topics: ["job.echo"]}
    decision: allow
"""

ROUTING = PoolRouting(
    topics={"job.default": ["default"], "job.echo": ["default"],
            "job.prod.deploy": ["default"]},
    pools={"default": PoolProfile()},
)


@pytest.fixture
def gpu_node():
    n = Node(clock=ManualClock(), routing=ROUTING, policy_yaml=POLICY,
             dispatch="device", device="cuda:0", backend="ext").start()
    n.add_device_worker_pool(n_workers=8, topics=["job.default", "job.echo"])
    return n


def test_native_ext_is_the_backend(gpu_node):
    from cordum_amd.ops import get_ext

    assert gpu_node._ext is get_ext(required=True)
    assert gpu_node.device_gate.device_active  # policy compiled to device tensors


def test_api_submitted_jobs_run_on_gpu(gpu_node):
    """gateway -> device engine -> K1/K2 -> device echo pool -> result."""
    from fastapi.testclient import TestClient

    from cordum_amd.gateway.app import create_app

    c = TestClient(create_app(gpu_node))
    c.headers.update({"X-API-Key": "test-key", "X-Principal-Role": "admin"})
    r = c.post("/api/v1/jobs", json={"prompt": "hello mi355x", "topic": "job.echo"})
    assert r.status_code == 200, r.text
    job_id = r.json()["job_id"]
    d = c.get(f"/api/v1/jobs/{job_id}").json()
    assert d["state"] == "SUCCEEDED"
    assert "hello mi355x" in str(d.get("result", ""))
    assert gpu_node.device_gate.jobs_evaluated >= 1
    assert gpu_node.scheduler.device_routed >= 1
    assert gpu_node.device_pools[0].jobs_executed >= 1


def test_bulk_submit_batch_through_kernels(gpu_node):
    n = gpu_node
    for i in range(2048):
        n.submit_job(JobRequest(job_id=f"b{i}", topic="job.echo", tenant_id="default"),
                     context=b'{"i": %d}' % i)
    n.drain()
    states = [n.job_store.get_state(f"b{i}") for i in range(2048)]
    assert all(s == JobState.SUCCEEDED for s in states)
    # one flush = one K1 launch + one K2 launch for the whole batch
    assert n.device_gate.batches_evaluated <= 4
    assert n.device_pools[0].batches_executed <= 4
    assert n.scheduler.device_routed == 2048


def test_decisions_match_host_oracle_on_gpu():
    host = Node(clock=ManualClock(), routing=ROUTING, policy_yaml=POLICY).start()
    dev = Node(clock=ManualClock(), routing=ROUTING, policy_yaml=POLICY,
               dispatch="device", device="cuda:0", backend="ext").start()
    for n in (host, dev):
        n.add_worker("w0", topics=["job.default", "job.echo", "job.prod.deploy"])
    cases = []
    for i in range(256):
        topic = ["job.echo", "job.default", "job.prod.deploy"][i % 3]
        meta = JobMetadata(risk_tags=["dangerous"] if i % 5 == 0 else [])
        cases.append(JobRequest(job_id=f"c{i}", topic=topic, tenant_id="default", meta=meta))
    for n in (host, dev):
        for req in cases:
            n.submit_job(JobRequest.decode(req.encode()), context=b"{}")
        n.drain()
    for req in cases:
        assert host.job_store.get_state(req.job_id) == dev.job_store.get_state(req.job_id)
        rh = host.job_store.get_safety_decision(req.job_id)
        rd = dev.job_store.get_safety_decision(req.job_id)
        if rh is not None:
            assert (rh.decision, rh.rule_id) == (rd.decision, rd.rule_id)
