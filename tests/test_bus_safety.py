"""Bus semantics (queue groups, wildcards, dedup, NAK-delay) and safety
policy/kernel evaluation matrix (oracle: safetykernel/kernel_test.go,
safety_policy.go tests)."""
import pytest

from cordum_amd.bus import LoopbackBus, RetryAfter, compute_msg_id, subject_matches
from cordum_amd.protocol.capv2 import (
    BusPacket,
    DecisionType,
    Heartbeat,
    JobMetadata,
    JobRequest,
    PolicyCheckRequest,
)
from cordum_amd.safety import (
    AllowAllSafety,
    PolicyInput,
    SafetyKernel,
    parse_safety_policy,
)
from cordum_amd.store import ConfigService
from cordum_amd.utils.clock import ManualClock


# --- subject matching --------------------------------------------------------


@pytest.mark.parametrize(
    "pattern,subject,want",
    [
        ("sys.job.submit", "sys.job.submit", True),
        ("sys.job.>", "sys.job.submit", True),
        ("sys.job.>", "sys.job.result.extra", True),
        ("sys.job.>", "sys.heartbeat", False),
        ("sys.*", "sys.heartbeat", True),
        ("sys.*", "sys.job.submit", False),
        ("worker.w1.jobs", "worker.w1.jobs", True),
        ("worker.*.jobs", "worker.w2.jobs", True),
        (">", "anything.at.all", True),
    ],
)
def test_subject_matches(pattern, subject, want):
    assert subject_matches(pattern, subject) is want


# --- loopback bus ------------------------------------------------------------


def test_queue_group_delivers_to_one_member():
    bus = LoopbackBus(clock=ManualClock())
    got = {"a": 0, "b": 0, "plain": 0}
    bus.subscribe("sys.job.submit", lambda s, p: got.__setitem__("a", got["a"] + 1), queue_group="g")
    bus.subscribe("sys.job.submit", lambda s, p: got.__setitem__("b", got["b"] + 1), queue_group="g")
    bus.subscribe("sys.job.>", lambda s, p: got.__setitem__("plain", got["plain"] + 1))
    for i in range(4):
        pkt = BusPacket(job_request=JobRequest(job_id=f"j{i}", topic="job.x"))
        bus.publish("sys.job.submit", pkt)
    assert got["a"] + got["b"] == 4
    assert got["a"] == 2 and got["b"] == 2  # round-robin
    assert got["plain"] == 4  # broadcast tap sees all


def test_msg_id_dedup_on_durable_subject():
    clock = ManualClock()
    bus = LoopbackBus(clock=clock)
    seen = []
    bus.subscribe("sys.job.submit", lambda s, p: seen.append(p.job_id()))
    pkt = BusPacket(job_request=JobRequest(job_id="j1", topic="job.x"))
    bus.publish("sys.job.submit", pkt)
    bus.publish("sys.job.submit", pkt)  # duplicate within window
    assert seen == ["j1"]
    clock.advance(121)  # outside the 2-minute window
    bus.publish("sys.job.submit", pkt)
    assert seen == ["j1", "j1"]


def test_msg_id_override_label_allows_republish():
    bus = LoopbackBus(clock=ManualClock())
    seen = []
    bus.subscribe("sys.job.submit", lambda s, p: seen.append(1))
    req = JobRequest(job_id="j1", topic="job.x")
    bus.publish("sys.job.submit", BusPacket(job_request=req))
    pkt2 = BusPacket(job_request=req, labels={"cordum.bus_msg_id": "approval-republish-1"})
    bus.publish("sys.job.submit", pkt2)
    assert len(seen) == 2


def test_heartbeats_not_deduped():
    bus = LoopbackBus(clock=ManualClock())
    seen = []
    bus.subscribe("sys.heartbeat", lambda s, p: seen.append(1))
    hb = BusPacket(heartbeat=Heartbeat(worker_id="w1"))
    bus.publish("sys.heartbeat", hb)
    bus.publish("sys.heartbeat", hb)
    assert len(seen) == 2
    assert compute_msg_id("sys.heartbeat", hb) is None


def test_retry_after_nak_redelivery():
    clock = ManualClock()
    bus = LoopbackBus(clock=clock)
    attempts = []

    def handler(s, p):
        attempts.append(clock.now())
        if len(attempts) < 3:
            raise RetryAfter(5.0, "throttled")

    bus.subscribe("sys.job.submit", handler, queue_group="g")
    bus.publish("sys.job.submit", BusPacket(job_request=JobRequest(job_id="jr", topic="job.x")))
    assert len(attempts) == 1
    bus.pump()
    assert len(attempts) == 1  # not due yet
    clock.advance(5.1)
    bus.pump()
    assert len(attempts) == 2
    clock.advance(5.1)
    bus.pump()
    assert len(attempts) == 3
    clock.advance(60)
    assert bus.pump() == 0  # handler succeeded; nothing pending


def test_delayed_publish():
    clock = ManualClock()
    bus = LoopbackBus(clock=clock)
    seen = []
    bus.subscribe("sys.job.submit", lambda s, p: seen.append(p.job_id()))
    bus.publish_after(10, "sys.job.submit", BusPacket(job_request=JobRequest(job_id="d1", topic="job.x")))
    bus.pump()
    assert seen == []
    clock.advance(10.5)
    bus.pump()
    assert seen == ["d1"]


# --- safety policy evaluation -------------------------------------------------

POLICY_YAML = """
version: v1
default_tenant: default
rules:
  - id: deny-prod-deploy
    decision: deny
    reason: deploys are blocked
    match:
      topics: ["job.deploy.*"]
      tenants: [default]
  - id: approve-secrets
    decision: require_approval
    reason: secrets need review
    match:
      risk_tags: [secrets]
  - id: constrain-gpu
    decision: allow_with_constraints
    match:
      requires: [gpu]
    constraints:
      budgets:
        max_retries: 2
        max_runtime_ms: 60000
      redaction_level: strict
  - id: throttle-bulk
    decision: throttle
    reason: bulk lane
    match:
      labels: {lane: bulk}
  - id: approve-secret-flag
    decision: require_approval
    reason: secrets present
    match:
      secrets_present: true
tenants:
  legacy-only:
    allow_topics: ["job.echo"]
    deny_topics: ["job.admin.*"]
"""


def make_kernel(cache_ttl=0.0, configsvc=None, clock=None):
    policy = parse_safety_policy(POLICY_YAML)
    return SafetyKernel(policy, configsvc=configsvc, cache_ttl_s=cache_ttl, clock=clock or ManualClock())


def req(topic="job.echo", tenant="default", **kw):
    r = PolicyCheckRequest(job_id="j1", topic=topic, tenant=tenant)
    for k, v in kw.items():
        setattr(r, k, v)
    return r


def test_topic_gate():
    k = make_kernel()
    assert k.check(req(topic="")).decision == DecisionType.DENY
    assert k.check(req(topic="sys.job.submit")).decision == DecisionType.DENY
    assert k.check(req(topic="job.echo")).decision == DecisionType.ALLOW


def test_first_match_wins_and_reason():
    k = make_kernel()
    r = k.check(req(topic="job.deploy.prod"))
    assert r.decision == DecisionType.DENY
    assert r.rule_id == "deny-prod-deploy"
    assert r.reason == "deploys are blocked"


def test_require_approval_sets_ref():
    k = make_kernel()
    r = k.check(req(meta=JobMetadata(risk_tags=["secrets"])))
    assert r.decision == DecisionType.REQUIRE_HUMAN
    assert r.approval_required and r.approval_ref == "j1"


def test_constraints_decision():
    k = make_kernel()
    r = k.check(req(meta=JobMetadata(requires=["gpu"])))
    assert r.decision == DecisionType.ALLOW_WITH_CONSTRAINTS
    assert r.constraints.budgets.max_retries == 2
    assert r.constraints.redaction_level == "strict"


def test_label_match_throttle():
    k = make_kernel()
    r = k.check(req(labels={"lane": "bulk"}))
    assert r.decision == DecisionType.THROTTLE


def test_secrets_label_triggers_approval():
    k = make_kernel()
    r = k.check(req(labels={"secrets_present": "true"}))
    assert r.decision == DecisionType.REQUIRE_HUMAN


def test_legacy_tenant_rules():
    yaml_text = """
version: v1
tenants:
  t1:
    allow_topics: ["job.echo"]
    deny_topics: ["job.admin.*"]
"""
    k = SafetyKernel(parse_safety_policy(yaml_text), cache_ttl_s=0.0)
    assert k.check(req(topic="job.admin.users", tenant="t1")).decision == DecisionType.DENY
    assert k.check(req(topic="job.echo", tenant="t1")).decision == DecisionType.ALLOW


def test_effective_config_restricts_topics():
    k = make_kernel()
    eff = b'{"safety": {"allowed_topics": ["job.echo"]}}'
    assert k.check(req(topic="job.other", effective_config=eff)).decision == DecisionType.DENY
    assert k.check(req(topic="job.echo", effective_config=eff)).decision == DecisionType.ALLOW
    eff2 = b'{"safety": {"denied_topics": ["job.echo"]}}'
    assert k.check(req(topic="job.echo", effective_config=eff2)).decision == DecisionType.DENY


def test_mcp_deny_via_labels():
    yaml_text = """
version: v1
tenants:
  default:
    mcp:
      deny_tools: ["shell"]
"""
    k = SafetyKernel(parse_safety_policy(yaml_text), cache_ttl_s=0.0)
    r = k.check(req(labels={"mcp.tool": "shell"}))
    assert r.decision == DecisionType.DENY
    assert "shell" in r.reason
    assert k.check(req(labels={"mcp.tool": "search"})).decision == DecisionType.ALLOW


def test_decision_cache_hits_and_snapshot_binding():
    clock = ManualClock()
    k = make_kernel(cache_ttl=30.0, clock=clock)
    r1 = k.check(req(meta=JobMetadata(risk_tags=["secrets"])))
    assert k.cache_size() == 1
    # same request different job_id -> cache hit, approval_ref rebound
    r2 = k.check(PolicyCheckRequest(job_id="j2", topic="job.echo", tenant="default", meta=JobMetadata(risk_tags=["secrets"])))
    assert r2.approval_ref == "j2"
    assert r2.policy_snapshot == r1.policy_snapshot
    clock.advance(31)
    k.check(req())  # expired entries tolerated


def test_bundle_merge_and_hot_reload():
    cs = ConfigService()
    clock = ManualClock()
    k = make_kernel(configsvc=cs, clock=clock)
    snap0 = k.snapshot
    assert k.check(req(topic="job.newthing")).decision == DecisionType.ALLOW
    cs.set("system", "policy", {
        "enabled": True,
        "bundles": {"extra": {"enabled": True, "content": "version: v2\nrules:\n  - id: deny-new\n    decision: deny\n    match:\n      topics: [\"job.newthing\"]\n"}},
    })
    assert k.snapshot != snap0
    r = k.check(req(topic="job.newthing"))
    assert r.decision == DecisionType.DENY and r.rule_id == "deny-new"
    assert snap0 in k.list_snapshots() and k.snapshot in k.list_snapshots()


def test_allow_all_seam():
    a = AllowAllSafety()
    assert a.check(req()).decision == DecisionType.ALLOW


def test_no_rule_matches_default_allow():
    p = parse_safety_policy(POLICY_YAML)
    d = p.evaluate(PolicyInput(tenant="default", topic="job.misc"))
    assert d.decision == "allow" and d.rule_id == ""
