"""Protocol-layer tests: state machine table, CAP v2 codec round-trip,
glob matching oracle, job hash stability (approval binding)."""
import pytest

from cordum_amd.protocol import JobState, can_transition, is_terminal, parse_state, transition_lut
from cordum_amd.protocol.capv2 import (
    Budget,
    BusPacket,
    DecisionType,
    Heartbeat,
    JobMetadata,
    JobPriority,
    JobRequest,
    JobResult,
    JobStatus,
    PolicyCheckResponse,
    PolicyConstraints,
    PolicyRemediation,
    BudgetConstraints,
)
from cordum_amd.utils import glob_match, job_hash, topic_matches
from cordum_amd.utils.canonical_json import canonical_json, canonical_json_hash


# --- state machine (oracle job_store.go:60-82) -----------------------------


def test_terminal_states():
    for s in (JobState.SUCCEEDED, JobState.FAILED, JobState.CANCELLED, JobState.TIMEOUT, JobState.DENIED):
        assert is_terminal(s)
        # no transitions out of terminal states
        for t in JobState:
            assert not can_transition(s, t)
    for s in (JobState.PENDING, JobState.SCHEDULED, JobState.DISPATCHED, JobState.RUNNING):
        assert not is_terminal(s)


def test_transition_table_matches_reference():
    assert can_transition(JobState.UNSPECIFIED, JobState.PENDING)
    assert can_transition(JobState.UNSPECIFIED, JobState.FAILED)
    assert not can_transition(JobState.UNSPECIFIED, JobState.SUCCEEDED)
    assert can_transition(JobState.PENDING, JobState.APPROVAL_REQUIRED)
    assert can_transition(JobState.APPROVAL_REQUIRED, JobState.PENDING)
    assert can_transition(JobState.SCHEDULED, JobState.SUCCEEDED)
    assert can_transition(JobState.SCHEDULED, JobState.CANCELLED)
    assert not can_transition(JobState.PENDING, JobState.SUCCEEDED)
    assert not can_transition(JobState.PENDING, JobState.CANCELLED)
    assert can_transition(JobState.DISPATCHED, JobState.CANCELLED)
    assert can_transition(JobState.RUNNING, JobState.TIMEOUT)
    assert not can_transition(JobState.RUNNING, JobState.PENDING)


def test_transition_lut_matches_table():
    lut = transition_lut()
    for f in JobState:
        for t in JobState:
            assert bool(lut[int(f)][int(t)]) == can_transition(f, t)


def test_parse_state():
    assert parse_state("PENDING") is JobState.PENDING
    assert parse_state("") is JobState.UNSPECIFIED
    assert str(JobState.APPROVAL_REQUIRED) == "APPROVAL_REQUIRED"
    with pytest.raises(ValueError):
        parse_state("BOGUS")


# --- CAP v2 codec -----------------------------------------------------------


def make_req(**kw):
    req = JobRequest(
        job_id="job-1",
        topic="job.default",
        priority=JobPriority.BATCH,
        tenant_id="default",
        context_ptr="redis://ctx:job-1",
        labels={"workflow_id": "wf1", "b": "2", "a": "1"},
        env={"tenant_id": "default"},
        meta=JobMetadata(actor_id="alice", risk_tags=["secrets"], requires=["gpu"]),
        budget=Budget(max_tokens=8000, deadline_ms=60_000),
    )
    for k, v in kw.items():
        setattr(req, k, v)
    return req


def test_jobrequest_roundtrip():
    req = make_req()
    blob = req.encode()
    back = JobRequest.decode(blob)
    assert back.job_id == "job-1"
    assert back.topic == "job.default"
    assert back.priority == JobPriority.BATCH
    assert back.labels == {"workflow_id": "wf1", "a": "1", "b": "2"}
    assert back.meta.risk_tags == ["secrets"]
    assert back.budget.deadline_ms == 60_000
    assert back.encode() == blob  # deterministic


def test_buspacket_payload():
    pkt = BusPacket(trace_id="t1", job_request=make_req())
    name, payload = pkt.payload()
    assert name == "job_request"
    back = BusPacket.decode(pkt.encode())
    assert back.trace_id == "t1"
    assert back.job_request.job_id == "job-1"
    assert back.job_id() == "job-1"

    res = BusPacket(job_result=JobResult(job_id="j2", status=JobStatus.SUCCEEDED))
    back2 = BusPacket.decode(res.encode())
    assert back2.job_result.status == JobStatus.SUCCEEDED


def test_json_roundtrip():
    req = make_req()
    d = req.to_dict()
    assert d["jobId"] == "job-1"
    assert d["priority"] == "JOB_PRIORITY_BATCH"
    back = JobRequest.from_dict(d)
    assert back.encode() == req.encode()


def test_heartbeat_roundtrip():
    hb = Heartbeat(worker_id="w1", pool="default", active_jobs=3, cpu_load=42.5,
                   capabilities=["echo", "llm"], max_parallel_jobs=8)
    back = Heartbeat.decode(hb.encode())
    assert back.worker_id == "w1"
    assert back.cpu_load == 42.5
    assert back.capabilities == ["echo", "llm"]


def test_policy_response_roundtrip():
    resp = PolicyCheckResponse(
        decision=DecisionType.ALLOW_WITH_CONSTRAINTS,
        rule_id="r1",
        policy_snapshot="v1:abc",
        constraints=PolicyConstraints(budgets=BudgetConstraints(max_retries=2), redaction_level="strict"),
        remediations=[PolicyRemediation(id="rem1", replacement_topic="job.safe")],
    )
    back = PolicyCheckResponse.decode(resp.encode())
    assert back.decision == DecisionType.ALLOW_WITH_CONSTRAINTS
    assert back.constraints.budgets.max_retries == 2
    assert back.remediations[0].replacement_topic == "job.safe"
    assert PolicyCheckResponse.from_dict(back.to_dict()).encode() == resp.encode()


# --- job hash (oracle job_hash.go:15-48) ------------------------------------


def test_job_hash_strips_approval_labels_and_effective_config():
    req = make_req()
    h0 = job_hash(req)

    approved = make_req()
    approved.labels = dict(approved.labels)
    approved.labels["approval_granted"] = "true"
    approved.labels["approval_reason"] = "ok"
    approved.labels["cordum.bus_msg_id"] = "fresh"
    approved.env = dict(approved.env)
    approved.env["CORDUM_EFFECTIVE_CONFIG"] = "{\"x\":1}"
    assert job_hash(approved) == h0

    changed = make_req(topic="job.other")
    assert job_hash(changed) != h0


# --- glob matching (oracle path.Match) --------------------------------------


@pytest.mark.parametrize(
    "pattern,name,want",
    [
        ("job.*", "job.default", True),
        ("job.*", "job.a.b", True),  # '.' is not a separator in path.Match
        ("job.*", "sys.job.submit", False),
        ("*", "anything", True),
        ("*", "a/b", False),  # '*' does not cross '/'
        ("a/*/c", "a/b/c", True),
        ("?ob.x", "job.x", True),
        ("[jk]ob.*", "job.z", True),
        ("[!j]ob.*", "job.z", False),
        ("[a-m]ob.*", "job.z", True),
        ("job.\\*", "job.*", True),
        ("job.\\*", "job.x", False),
    ],
)
def test_glob_match(pattern, name, want):
    assert glob_match(pattern, name) is want


def test_topic_matches_malformed_pattern_no_match():
    assert topic_matches("job.[", "job.x") is False
    assert topic_matches("job.x", "job.x") is True


# --- canonical json ---------------------------------------------------------


def test_canonical_json_stable():
    a = {"b": 1, "a": {"z": [1, 2], "y": "s"}}
    b = {"a": {"y": "s", "z": [1, 2]}, "b": 1}
    assert canonical_json(a) == canonical_json(b)
    assert canonical_json_hash(a) == canonical_json_hash(b)
    assert canonical_json({"k": "v"}) == '{"k":"v"}'
