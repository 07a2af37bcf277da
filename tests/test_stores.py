"""Store-layer tests, mirroring the reference's miniredis-backed suites
(memory/job_store_test.go, locks/redis_store_test.go, dlq/artifacts/configsvc)."""
import pytest

from cordum_amd.protocol import JobState
from cordum_amd.protocol.capv2 import JobRequest
from cordum_amd.store import (
    ArtifactStore,
    ConfigService,
    DLQEntry,
    DLQStore,
    InvalidTransition,
    JobStore,
    LockService,
    MemoryStore,
    SchemaRegistry,
    json_merge_patch,
)
from cordum_amd.store.memory_store import key_from_pointer, pointer_for_key
from cordum_amd.store.schema_registry import validate_value
from cordum_amd.utils.clock import ManualClock


@pytest.fixture
def clock():
    return ManualClock()


# --- job store ---------------------------------------------------------------


def test_job_store_transitions_and_indexes(clock):
    js = JobStore(clock=clock)
    js.set_tenant("j1", "acme")
    js.set_state("j1", JobState.PENDING)
    assert js.get_state("j1") == JobState.PENDING
    assert js.list_jobs_by_state(JobState.PENDING) == ["j1"]
    assert js.tenant_active_count("acme") == 1

    js.set_state("j1", JobState.SCHEDULED)
    assert js.get_job_meta("j1")["attempts"] == 1
    assert js.list_jobs_by_state(JobState.PENDING) == []
    js.set_state("j1", JobState.DISPATCHED)
    js.set_state("j1", JobState.RUNNING)
    js.set_state("j1", JobState.SUCCEEDED)
    assert js.tenant_active_count("acme") == 0
    events = js.get_events("j1")
    assert [e.split("|")[1] for e in events] == [
        "PENDING", "SCHEDULED", "DISPATCHED", "RUNNING", "SUCCEEDED",
    ]
    # terminal: no further transitions
    with pytest.raises(InvalidTransition):
        js.set_state("j1", JobState.RUNNING)


def test_job_store_invalid_transition(clock):
    js = JobStore(clock=clock)
    js.set_state("j2", JobState.PENDING)
    with pytest.raises(InvalidTransition):
        js.set_state("j2", JobState.SUCCEEDED)  # PENDING -> SUCCEEDED not allowed


def test_attempts_increment_only_on_entering_scheduled(clock):
    js = JobStore(clock=clock)
    js.set_state("j3", JobState.PENDING)
    js.set_state("j3", JobState.SCHEDULED)
    js.set_state("j3", JobState.DISPATCHED)
    js.set_state("j3", JobState.RUNNING)
    js.set_state("j3", JobState.TIMEOUT)
    assert js.get_job_meta("j3")["attempts"] == 1


def test_deadline_index(clock):
    js = JobStore(clock=clock)
    js.set_state("a", JobState.PENDING)
    js.set_deadline("a", clock.now_micros() + 5_000_000)
    assert js.list_expired_deadlines() == []
    clock.advance(6)
    assert js.list_expired_deadlines() == ["a"]
    # terminal state clears deadline
    js.set_state("a", JobState.FAILED)
    assert js.list_expired_deadlines() == []


def test_idempotency_scoped(clock):
    js = JobStore(clock=clock)
    ok, jid = js.try_set_idempotency_key("t1", "k", "job-a")
    assert ok and jid == "job-a"
    ok, jid = js.try_set_idempotency_key("t1", "k", "job-b")
    assert not ok and jid == "job-a"
    ok, _ = js.try_set_idempotency_key("t2", "k", "job-c")  # different tenant
    assert ok


def test_cancel_job(clock):
    js = JobStore(clock=clock)
    js.set_state("c1", JobState.PENDING)
    assert js.cancel_job("c1")
    assert js.get_state("c1") == JobState.CANCELLED
    assert not js.cancel_job("c1")  # already terminal
    assert not js.cancel_job("nope")


def test_job_request_roundtrip_and_snapshot(clock):
    js = JobStore(clock=clock)
    req = JobRequest(job_id="j9", topic="job.x", tenant_id="t")
    js.set_job_request("j9", req)
    js.set_tenant("j9", "t")
    js.set_state("j9", JobState.PENDING)
    js.add_job_to_trace("tr1", "j9")
    snap = js.snapshot()
    assert "job:meta:j9" in snap and "trace:tr1" in snap
    js2 = JobStore(clock=clock)
    js2.restore(snap)
    assert js2.get_state("j9") == JobState.PENDING
    assert js2.get_job_request("j9").topic == "job.x"
    assert js2.get_trace("tr1") == ["j9"]


def test_job_lock(clock):
    js = JobStore(clock=clock)
    with js.job_lock("j1", owner="s1") as ok:
        assert ok
        assert not js.try_lock("lock:job:j1", "s2")
    assert js.try_lock("lock:job:j1", "s2")


# --- memory store ------------------------------------------------------------


def test_memory_pointers(clock):
    ms = MemoryStore(clock=clock)
    ptr = ms.put_context("j1", b"hello")
    assert ptr == "redis://ctx:j1"
    assert key_from_pointer(ptr) == "ctx:j1"
    assert ms.get_pointer(ptr) == b"hello"
    clock.advance(25 * 3600)  # past 24h TTL
    assert ms.get_pointer(ptr) is None
    with pytest.raises(ValueError):
        key_from_pointer("bogus://x")
    assert pointer_for_key("res:a") == "redis://res:a"


# --- DLQ ---------------------------------------------------------------------


def test_dlq_cap_and_paging(clock):
    dlq = DLQStore(clock=clock, cap=5)
    for i in range(8):
        clock.advance(1)
        dlq.add(DLQEntry(job_id=f"j{i}", reason_code="no_workers"))
    assert len(dlq) == 5
    page, cursor = dlq.list(limit=2)
    assert [e.job_id for e in page] == ["j7", "j6"]
    page2, _ = dlq.list(limit=10, cursor=cursor)
    assert [e.job_id for e in page2] == ["j5", "j4", "j3"]
    assert dlq.delete("j7")
    assert not dlq.delete("j7")


# --- locks -------------------------------------------------------------------


def test_locks_shared_exclusive_upgrade(clock):
    ls = LockService(clock=clock)
    assert ls.acquire("r", "a", "shared")
    assert ls.acquire("r", "b", "shared")
    assert not ls.acquire("r", "c", "exclusive")  # two shared owners block exclusive
    assert ls.release("r", "b")
    assert ls.acquire("r", "a", "exclusive")  # sole owner upgrade
    assert not ls.acquire("r", "b", "shared")
    assert ls.release("r", "a") and ls.release("r", "a")
    assert ls.get("r") is None


def test_lock_ttl_expiry(clock):
    ls = LockService(clock=clock)
    assert ls.acquire("r", "a", "exclusive", ttl_s=10)
    clock.advance(11)
    assert ls.acquire("r", "b", "exclusive")
    assert ls.renew("r", "b", ttl_s=5)
    assert not ls.renew("r", "a")


# --- artifacts ---------------------------------------------------------------


def test_artifacts_retention(clock):
    ms = MemoryStore(clock=clock)
    arts = ArtifactStore(ms, clock=clock)
    ptr = arts.put(b"data", retention="short")
    assert ptr.startswith("redis://art:")
    got = arts.get_pointer(ptr)
    assert got is not None and got[0] == b"data" and got[1].retention == "short"
    clock.advance(25 * 3600)
    assert arts.get_pointer(ptr) is None  # short = 24h


# --- schema registry ---------------------------------------------------------


def test_schema_validation():
    reg = SchemaRegistry()
    reg.put("s1", {"type": "object", "required": ["name"], "properties": {"name": {"type": "string"}, "n": {"type": "integer", "minimum": 1}}})
    ok, errs = reg.validate_against("s1", {"name": "x", "n": 3})
    assert ok, errs
    ok, errs = reg.validate_against("s1", {"n": 0})
    assert not ok and len(errs) == 2
    ok, errs = reg.validate_against("missing", {})
    assert not ok
    assert validate_value({"type": "array", "items": {"type": "string"}}, ["a", 1])


# --- configsvc ---------------------------------------------------------------


def test_configsvc_overlay_merge():
    cs = ConfigService()
    cs.set("system", "default", {"safety": {"x": 1, "y": 1}, "retry": {"max": 3}})
    cs.set("org", "acme", {"safety": {"y": 2}})
    cs.set("workflow", "wf1", {"retry": {"max": 5}})
    snap = cs.effective(org="acme", workflow="wf1")
    assert snap.config["safety"] == {"x": 1, "y": 2}
    assert snap.config["retry"] == {"max": 5}
    assert "sys:1" in snap.version and "org:1" in snap.version
    h1 = snap.hash
    cs.set("org", "acme", {"safety": {"y": 3}})
    assert cs.effective(org="acme", workflow="wf1").hash != h1


def test_configsvc_watch_and_patch():
    cs = ConfigService()
    fired = []
    cs.watch(lambda s, i: fired.append((s, i)))
    cs.set("system", "policy", {"bundles": {"b1": {"content": "x"}}})
    assert fired == [("system", "policy")]
    cs.patch("system", "policy", {"bundles": {"b2": {"content": "y"}, "b1": None}})
    doc = cs.get("system", "policy")
    assert "b2" in doc["bundles"] and "b1" not in doc["bundles"]


def test_json_merge_patch():
    assert json_merge_patch({"a": {"b": 1}}, {"a": {"c": 2}}) == {"a": {"b": 1, "c": 2}}
    assert json_merge_patch({"a": 1}, {"a": None}) == {}
    assert json_merge_patch({"a": 1}, "str") == "str"


# --- pagination / cursors ----------------------------------------------------


def test_job_store_recent_pagination(clock):
    js = JobStore(clock=clock)
    for i in range(10):
        clock.advance(1)
        js.set_state(f"p{i}", JobState.PENDING)
    page1, cursor = js.list_recent(limit=4)
    assert page1 == [f"p{i}" for i in range(9, 5, -1)]
    assert cursor is not None
    page2, cursor2 = js.list_recent(limit=4, cursor=cursor)
    assert page2 == [f"p{i}" for i in range(5, 1, -1)]
    page3, cursor3 = js.list_recent(limit=4, cursor=cursor2)
    assert page3 == ["p1", "p0"] and cursor3 is None


def test_workflow_store_run_filters_and_pagination(clock):
    from cordum_amd.workflow import Workflow, WorkflowRun, WorkflowStore

    ws = WorkflowStore(clock=clock)
    for i in range(6):
        clock.advance(1)
        ws.create_run(WorkflowRun(id=f"r{i}", workflow_id="wfA" if i % 2 else "wfB",
                                  org_id="org", status="running" if i < 4 else "succeeded"))
    runs, cursor = ws.list_runs(limit=3)
    assert [r.id for r in runs] == ["r5", "r4", "r3"]
    runs2, _ = ws.list_runs(limit=10, cursor=cursor)
    assert [r.id for r in runs2] == ["r2", "r1", "r0"]
    only_a, _ = ws.list_runs(workflow_id="wfA")
    assert all(r.workflow_id == "wfA" for r in only_a)
    running, _ = ws.list_runs(status="running")
    assert len(running) == 4
    assert ws.count_active_runs("org") == 4
