"""Safety policy model + first-match evaluator (host oracle).

Semantics oracle: core/infra/config/safety_policy.go —
types :13-146, Evaluate :187-206 (first-match wins, default allow),
normalizeDecision :208-223 (unknown -> allow), legacyRules :225-257
(tenant deny topics then allow topics compiled to rules), matchRule :259-294
(tenants/topics/capabilities any-of; requires all-of; labels exact;
secrets_present tri-state; case-insensitive trimmed string compare,
empty value never matches), topic globs via path.Match :347-363,
MCP allow/deny :365-416.

This host evaluator is the oracle for the batched HIP policy kernel
(ops/hip/cordum_kernels.hip K1): the compiler in ops/policy_compile.py
lowers these rules to int tensors and the kernel must agree with
`Policy.evaluate` on every input (tests/test_policy_compile.py,
tests/test_gpu_kernels.py).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

from ..utils.globmatch import topic_matches as _path_match_topic
from ..protocol.capv2 import (
    BudgetConstraints,
    DiffConstraints,
    PolicyConstraints,
    PolicyRemediation,
    SandboxProfile,
    ToolchainConstraints,
)

DECISION_ALLOW = "allow"
DECISION_DENY = "deny"
DECISION_REQUIRE_APPROVAL = "require_approval"
DECISION_ALLOW_WITH_CONSTRAINTS = "allow_with_constraints"
DECISION_THROTTLE = "throttle"


def normalize_decision(raw: str) -> str:
    s = (raw or "").strip().lower()
    if s in ("allow", "permit"):
        return DECISION_ALLOW
    if s in ("deny", "block"):
        return DECISION_DENY
    if s in ("require_approval", "require-approval", "require_human"):
        return DECISION_REQUIRE_APPROVAL
    if s in ("allow_with_constraints", "allow-with-constraints"):
        return DECISION_ALLOW_WITH_CONSTRAINTS
    if s == "throttle":
        return DECISION_THROTTLE
    return DECISION_ALLOW


@dataclass
class MCPPolicy:
    allow_servers: List[str] = field(default_factory=list)
    deny_servers: List[str] = field(default_factory=list)
    allow_tools: List[str] = field(default_factory=list)
    deny_tools: List[str] = field(default_factory=list)
    allow_resources: List[str] = field(default_factory=list)
    deny_resources: List[str] = field(default_factory=list)
    allow_actions: List[str] = field(default_factory=list)
    deny_actions: List[str] = field(default_factory=list)

    @classmethod
    def from_dict(cls, d: Optional[Dict[str, Any]]) -> "MCPPolicy":
        d = d or {}
        return cls(**{k: list(d.get(k, []) or []) for k in cls.__dataclass_fields__})

    def is_empty(self) -> bool:
        return not any(getattr(self, k) for k in self.__dataclass_fields__)


@dataclass
class MCPRequest:
    server: str = ""
    tool: str = ""
    resource: str = ""
    action: str = ""

    def used(self) -> bool:
        return bool(self.server.strip() or self.tool.strip() or self.resource.strip() or self.action.strip())


@dataclass
class PolicyMatch:
    tenants: List[str] = field(default_factory=list)
    topics: List[str] = field(default_factory=list)
    capabilities: List[str] = field(default_factory=list)
    risk_tags: List[str] = field(default_factory=list)
    requires: List[str] = field(default_factory=list)
    pack_ids: List[str] = field(default_factory=list)
    actor_ids: List[str] = field(default_factory=list)
    actor_types: List[str] = field(default_factory=list)
    labels: Dict[str, str] = field(default_factory=dict)
    secrets_present: Optional[bool] = None
    mcp: MCPPolicy = field(default_factory=MCPPolicy)


@dataclass
class PolicyRule:
    id: str = ""
    match: PolicyMatch = field(default_factory=PolicyMatch)
    decision: str = DECISION_ALLOW
    reason: str = ""
    constraints: Optional[PolicyConstraints] = None
    remediations: List[PolicyRemediation] = field(default_factory=list)


@dataclass
class TenantPolicy:
    allow_topics: List[str] = field(default_factory=list)
    deny_topics: List[str] = field(default_factory=list)
    max_concurrent_jobs: int = 0
    mcp: MCPPolicy = field(default_factory=MCPPolicy)


@dataclass
class PolicyInput:
    tenant: str = ""
    topic: str = ""
    labels: Dict[str, str] = field(default_factory=dict)
    actor_id: str = ""
    actor_type: str = ""
    capability: str = ""
    risk_tags: List[str] = field(default_factory=list)
    requires: List[str] = field(default_factory=list)
    pack_id: str = ""
    secrets_present: bool = False
    mcp: MCPRequest = field(default_factory=MCPRequest)


@dataclass
class PolicyDecision:
    decision: str = DECISION_ALLOW
    reason: str = ""
    rule_id: str = ""
    constraints: Optional[PolicyConstraints] = None
    approval_required: bool = False
    remediations: List[PolicyRemediation] = field(default_factory=list)


# -- string matching helpers (safety_policy.go:294-345) ----------------------


def _contains(lst: List[str], value: str) -> bool:
    if not value:
        return False
    v = value.strip().lower()
    return any(x.strip().lower() == v for x in lst)


def _contains_any(lst: List[str], values: List[str]) -> bool:
    if not lst or not values:
        return False
    return any(_contains(lst, v) for v in values)


def _contains_all(values: List[str], required: List[str]) -> bool:
    return all(_contains(values, r) for r in required)


def match_topic(pattern: str, topic: str) -> bool:
    pattern = pattern.strip()
    if not pattern:
        return False
    return _path_match_topic(pattern, topic)


def mcp_allowed(policy: MCPPolicy, req: MCPRequest):
    """MCPAllowed (safety_policy.go:388-416): deny wins; allowlist if non-empty."""
    if not req.used():
        return True, ""
    checks = [
        ("server", req.server, policy.allow_servers, policy.deny_servers),
        ("tool", req.tool, policy.allow_tools, policy.deny_tools),
        ("resource", req.resource, policy.allow_resources, policy.deny_resources),
        ("action", req.action, policy.allow_actions, policy.deny_actions),
    ]
    for name, value, allow, deny in checks:
        if _contains(deny, value):
            return False, f'mcp {name} "{value}" denied'
        if allow and not _contains(allow, value):
            return False, f'mcp {name} "{value}" not allowed'
    return True, ""


def match_rule(match: PolicyMatch, inp: PolicyInput) -> bool:
    if match.tenants and not _contains(match.tenants, inp.tenant):
        return False
    if match.topics and not any(match_topic(p, inp.topic) for p in match.topics):
        return False
    if match.capabilities and not _contains(match.capabilities, inp.capability):
        return False
    if match.risk_tags and not _contains_any(match.risk_tags, inp.risk_tags):
        return False
    if match.requires and not _contains_all(inp.requires, match.requires):
        return False
    if match.pack_ids and not _contains(match.pack_ids, inp.pack_id):
        return False
    if match.actor_ids and not _contains(match.actor_ids, inp.actor_id):
        return False
    if match.actor_types and not _contains(match.actor_types, inp.actor_type):
        return False
    if match.secrets_present is not None and inp.secrets_present != match.secrets_present:
        return False
    if match.labels:
        if not inp.labels:
            return False
        for k, v in match.labels.items():
            if inp.labels.get(k, "") != v:
                return False
    ok, _ = mcp_allowed(match.mcp, inp.mcp)
    return ok


@dataclass
class SafetyPolicy:
    version: str = ""
    rules: List[PolicyRule] = field(default_factory=list)
    default_tenant: str = ""
    tenants: Dict[str, TenantPolicy] = field(default_factory=dict)

    def effective_rules(self) -> List[PolicyRule]:
        if self.rules:
            return self.rules
        return legacy_rules(self)

    def evaluate(self, inp: PolicyInput) -> PolicyDecision:
        for rule in self.effective_rules():
            if match_rule(rule.match, inp):
                return decision_from_rule(rule)
        return PolicyDecision(decision=DECISION_ALLOW)

    def explain(self, inp: PolicyInput) -> List[Dict[str, Any]]:
        """Full per-rule match row (the kernel's Explain mode)."""
        rows = []
        for rule in self.effective_rules():
            rows.append({"rule_id": rule.id, "matched": match_rule(rule.match, inp), "decision": normalize_decision(rule.decision)})
        return rows


def decision_from_rule(rule: PolicyRule) -> PolicyDecision:
    """PolicyDecision for a matched rule — the evaluate() match arm, shared
    with the batched device path so a K1 first-match index yields the exact
    same decision record (safety_policy.go:187-206)."""
    decision = normalize_decision(rule.decision)
    return PolicyDecision(
        decision=decision,
        reason=rule.reason,
        rule_id=rule.id,
        constraints=rule.constraints,
        approval_required=decision == DECISION_REQUIRE_APPROVAL,
        remediations=list(rule.remediations),
    )


def legacy_rules(policy: SafetyPolicy) -> List[PolicyRule]:
    """safety_policy.go:225-257; tenants iterated in sorted order for determinism
    (the Go map iteration order is unspecified — sorted is a superset guarantee)."""
    out: List[PolicyRule] = []
    for tenant in sorted(policy.tenants):
        tp = policy.tenants[tenant]
        for idx, pat in enumerate(tp.deny_topics):
            out.append(
                PolicyRule(
                    id=f"legacy:{tenant}:deny:{idx + 1}",
                    decision=DECISION_DENY,
                    reason=f'topic "{pat}" denied by tenant policy',
                    match=PolicyMatch(tenants=[tenant], topics=[pat], mcp=tp.mcp),
                )
            )
        for idx, pat in enumerate(tp.allow_topics):
            out.append(
                PolicyRule(
                    id=f"legacy:{tenant}:allow:{idx + 1}",
                    decision=DECISION_ALLOW,
                    match=PolicyMatch(tenants=[tenant], topics=[pat], mcp=tp.mcp),
                )
            )
    return out


# -- YAML parsing -------------------------------------------------------------


def _constraints_from_dict(d: Optional[Dict[str, Any]]) -> Optional[PolicyConstraints]:
    if not d:
        return None
    budgets = d.get("budgets") or {}
    sandbox = d.get("sandbox") or {}
    toolchain = d.get("toolchain") or {}
    diff = d.get("diff") or {}
    pc = PolicyConstraints(
        budgets=BudgetConstraints(
            max_runtime_ms=int(budgets.get("max_runtime_ms", 0) or 0),
            max_retries=int(budgets.get("max_retries", 0) or 0),
            max_artifact_bytes=int(budgets.get("max_artifact_bytes", 0) or 0),
            max_concurrent_jobs=int(budgets.get("max_concurrent_jobs", 0) or 0),
        )
        if budgets
        else None,
        sandbox=SandboxProfile(
            isolated=bool(sandbox.get("isolated", False)),
            network_allowlist=list(sandbox.get("network_allowlist", []) or []),
            filesystem="ro" if sandbox.get("fs_read_only") else ("rw" if sandbox.get("fs_read_write") else ""),
        )
        if sandbox
        else None,
        toolchain=ToolchainConstraints(
            allowed_tools=list(toolchain.get("allowed_tools", []) or []),
            allowed_commands=list(toolchain.get("allowed_commands", []) or []),
        )
        if toolchain
        else None,
        diff=DiffConstraints(
            max_files=int(diff.get("max_files", 0) or 0),
            max_lines=int(diff.get("max_lines", 0) or 0),
            deny_paths=list(diff.get("deny_path_globs", []) or []),
        )
        if diff
        else None,
        redaction_level=str(d.get("redaction_level", "") or ""),
    )
    if pc.budgets is None and pc.sandbox is None and pc.toolchain is None and pc.diff is None and not pc.redaction_level:
        return None
    return pc


def _remediations_from_list(lst) -> List[PolicyRemediation]:
    out = []
    for r in lst or []:
        out.append(
            PolicyRemediation(
                id=str(r.get("id", "") or ""),
                description=str(r.get("title", "") or r.get("summary", "") or ""),
                replacement_topic=str(r.get("replacement_topic", "") or ""),
                replacement_capability=str(r.get("replacement_capability", "") or ""),
                add_labels=dict(r.get("add_labels", {}) or {}),
                remove_labels=list(r.get("remove_labels", []) or []),
            )
        )
    return out


def parse_safety_policy(data: str) -> Optional[SafetyPolicy]:
    """ParseSafetyPolicy (safety_policy.go:169-185): empty -> None (allow-all)."""
    if not data or not data.strip():
        return None
    doc = yaml.safe_load(data)
    if doc is None:
        return None
    if not isinstance(doc, dict):
        raise ValueError("safety policy must be a mapping")
    return policy_from_dict(doc)


def policy_from_dict(doc: Dict[str, Any]) -> SafetyPolicy:
    rules = []
    for rd in doc.get("rules", []) or []:
        md = rd.get("match", {}) or {}
        sp = md.get("secrets_present", None)
        rules.append(
            PolicyRule(
                id=str(rd.get("id", "") or ""),
                decision=str(rd.get("decision", "") or ""),
                reason=str(rd.get("reason", "") or ""),
                match=PolicyMatch(
                    tenants=list(md.get("tenants", []) or []),
                    topics=list(md.get("topics", []) or []),
                    capabilities=list(md.get("capabilities", []) or []),
                    risk_tags=list(md.get("risk_tags", []) or []),
                    requires=list(md.get("requires", []) or []),
                    pack_ids=list(md.get("pack_ids", []) or []),
                    actor_ids=list(md.get("actor_ids", []) or []),
                    actor_types=list(md.get("actor_types", []) or []),
                    labels=dict(md.get("labels", {}) or {}),
                    secrets_present=bool(sp) if sp is not None else None,
                    mcp=MCPPolicy.from_dict(md.get("mcp")),
                ),
                constraints=_constraints_from_dict(rd.get("constraints")),
                remediations=_remediations_from_list(rd.get("remediations")),
            )
        )
    tenants = {}
    for name, td in (doc.get("tenants", {}) or {}).items():
        td = td or {}
        tenants[name] = TenantPolicy(
            allow_topics=list(td.get("allow_topics", []) or []),
            deny_topics=list(td.get("deny_topics", []) or []),
            max_concurrent_jobs=int(td.get("max_concurrent_jobs", 0) or 0),
            mcp=MCPPolicy.from_dict(td.get("mcp")),
        )
    return SafetyPolicy(
        version=str(doc.get("version", "") or ""),
        rules=rules,
        default_tenant=str(doc.get("default_tenant", "") or ""),
        tenants=tenants,
    )
