"""Safety kernel service — policy gate with snapshots, cache, hot reload.

Oracle: core/controlplane/safetykernel/kernel.go — evaluate :129-257 (topic
must start `job.`; tenant fallback default_tenant -> "default"; tenant-level
MCP gate after rule eval; effective-config topic/MCP restrictions), decision
cache keyed by deterministic request hash + snapshot :259-303 (job_id stripped
from the cache key; approval_ref re-bound per request), SHA-256 snapshot id
`version:hash|cfg:hash` :485-523, bundle merge from the config service's
`cfg:system:policy` doc :590-711, 30s watch loop -> native: synchronous
configsvc watch callback (no polling needed in-process).

In the reference this is a separate gRPC microservice; here it is fused into
the scheduler's process (the gRPC hop becomes a function
call). The BATCHED evaluation path for the device data plane lives in
ops/pipeline.py (K1 in ops/hip/cordum_kernels.hip) and is fed by the same
compiled policy (ops/policy_compile.py).
"""
from __future__ import annotations

import hashlib
import threading
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from ..protocol.capv2 import (
    DecisionType,
    PolicyCheckRequest,
    PolicyCheckResponse,
        PolicyRemediation,
)
from ..store.configsvc import ConfigService
from ..utils.clock import Clock, SYSTEM_CLOCK
from . import policy as pol

DEFAULT_CACHE_TTL_S = 30.0

POLICY_CFG_SCOPE = "system"
POLICY_CFG_ID = "policy"


def _pick_label(labels: Dict[str, str], *names: str) -> str:
    for n in names:
        v = (labels.get(n) or "").strip()
        if v:
            return v
    return ""


def extract_mcp_request(labels: Dict[str, str]) -> pol.MCPRequest:
    """kernel.go:395-404."""
    if not labels:
        return pol.MCPRequest()
    return pol.MCPRequest(
        server=_pick_label(labels, "mcp.server", "mcp_server", "mcpServer"),
        tool=_pick_label(labels, "mcp.tool", "mcp_tool", "mcpTool"),
        resource=_pick_label(labels, "mcp.resource", "mcp_resource", "mcpResource"),
        action=_pick_label(labels, "mcp.action", "mcp_action", "mcpAction").lower(),
    )


def secrets_present(risk_tags: List[str], labels: Dict[str, str]) -> bool:
    """kernel.go:381-393."""
    v = (labels or {}).get("secrets_present", "").strip()
    if v:
        return v in ("true", "1") or v.lower() == "yes"
    return any(t.lower() == "secrets" for t in risk_tags or [])


def parse_effective_safety(payload: bytes) -> Optional[Dict[str, Any]]:
    """ParseEffectiveSafety (infra/config/effective.go:12-40): the `safety`
    (or `data.safety`) section of the effective config JSON."""
    import json

    if not payload:
        return None
    try:
        top = json.loads(payload.decode("utf-8"))
    except Exception:
        return None
    if not isinstance(top, dict):
        return None
    raw = top.get("safety")
    if isinstance(raw, dict):
        return raw
    data = top.get("data")
    if isinstance(data, dict) and isinstance(data.get("safety"), dict):
        return data["safety"]
    return None


@dataclass
class _CacheEntry:
    resp: PolicyCheckResponse
    expires: float


class SafetyKernel:
    """In-process safety kernel. Thread-safe; hot-swaps policy on bundle writes."""

    def __init__(
        self,
        base_policy: Optional[pol.SafetyPolicy] = None,
        configsvc: Optional[ConfigService] = None,
        cache_ttl_s: float = DEFAULT_CACHE_TTL_S,
        clock: Clock = SYSTEM_CLOCK,
    ):
        self._clock = clock
        self._cache_ttl = cache_ttl_s
        self._mu = threading.RLock()
        self._base_policy = base_policy
        self._base_yaml_hash = ""
        self._configsvc = configsvc
        self._policy: Optional[pol.SafetyPolicy] = None
        self._snapshot = ""
        self._snapshots: List[str] = []
        self._cache: Dict[str, _CacheEntry] = {}
        self._compiled_listener = None  # device policy recompile hook
        self.reload()
        if configsvc is not None:
            configsvc.watch(self._on_cfg_write)

    # -- policy loading / merging ---------------------------------------------
    def _on_cfg_write(self, scope: str, doc_id: str) -> None:
        if scope == POLICY_CFG_SCOPE and doc_id == POLICY_CFG_ID:
            self.reload()

    def set_base_policy(self, policy: Optional[pol.SafetyPolicy], yaml_text: str = "") -> None:
        with self._mu:
            self._base_policy = policy
            self._base_yaml_hash = hashlib.sha256(yaml_text.encode()).hexdigest()[:16] if yaml_text else ""
        self.reload()

    def reload(self) -> str:
        """Merge base policy with enabled config-service bundle fragments,
        recompute the snapshot id (kernel.go:523-711)."""
        with self._mu:
            merged = _clone_policy(self._base_policy) if self._base_policy else pol.SafetyPolicy()
            cfg_hash = ""
            enabled = True
            if self._configsvc is not None:
                doc = self._configsvc.get(POLICY_CFG_SCOPE, POLICY_CFG_ID) or {}
                enabled = doc.get("enabled", True)
                bundles = doc.get("bundles", {}) or {}
                if enabled:
                    for bundle_id in sorted(bundles):
                        b = bundles[bundle_id] or {}
                        if not b.get("enabled", True):
                            continue
                        content = b.get("content", "")
                        try:
                            frag = pol.parse_safety_policy(content)
                        except Exception:
                            continue
                        if frag is None:
                            continue
                        merged.rules.extend(frag.rules)
                        for t, tp in frag.tenants.items():
                            merged.tenants[t] = tp
                        if frag.default_tenant:
                            merged.default_tenant = frag.default_tenant
                h = hashlib.sha256()
                for bundle_id in sorted(bundles):
                    b = bundles[bundle_id] or {}
                    h.update(bundle_id.encode())
                    h.update(str(b.get("enabled", True)).encode())
                    h.update((b.get("content", "") or "").encode())
                cfg_hash = h.hexdigest()[:16]
            version = merged.version or "v0"
            base_part = f"{version}:{self._base_yaml_hash or _policy_hash(self._base_policy)}"
            snapshot = f"{base_part}|cfg:{cfg_hash or '0'}"
            self._policy = merged if enabled else _clone_policy(self._base_policy)
            if snapshot != self._snapshot:
                self._snapshot = snapshot
                self._snapshots.append(snapshot)
                self._snapshots = self._snapshots[-32:]
                self._cache.clear()
            listener = self._compiled_listener
            policy_now = self._policy
        if listener is not None:
            listener(policy_now, snapshot)
        return snapshot

    def on_policy_swap(self, callback) -> None:
        """Register a hook fired after each reload with (policy, snapshot) —
        the device policy-table compiler subscribes here."""
        self._compiled_listener = callback
        with self._mu:
            callback(self._policy, self._snapshot)

    @property
    def snapshot(self) -> str:
        with self._mu:
            return self._snapshot

    def list_snapshots(self) -> List[str]:
        with self._mu:
            return list(self._snapshots)

    def current_policy(self) -> Optional[pol.SafetyPolicy]:
        with self._mu:
            return self._policy

    # -- evaluation -------------------------------------------------------------
    def check(self, req: PolicyCheckRequest) -> PolicyCheckResponse:
        return self.evaluate(req)

    def simulate(self, req: PolicyCheckRequest) -> PolicyCheckResponse:
        return self.evaluate(req)

    def explain(self, req: PolicyCheckRequest) -> PolicyCheckResponse:
        return self.evaluate(req)

    def explain_rows(self, req: PolicyCheckRequest) -> List[Dict[str, Any]]:
        inp = self.input_from_request(req)
        with self._mu:
            policy = self._policy
        return policy.explain(inp) if policy else []

    def input_from_request(self, req: PolicyCheckRequest) -> pol.PolicyInput:
        meta = req.meta
        tenant = (req.tenant or "").strip()
        with self._mu:
            policy = self._policy
        if not tenant and policy is not None:
            tenant = (policy.default_tenant or "").strip()
        if not tenant:
            tenant = "default"
        risk_tags = list(meta.risk_tags) if meta else []
        return pol.PolicyInput(
            tenant=tenant,
            topic=(req.topic or "").strip(),
            labels=dict(req.labels),
            actor_id=meta.actor_id if meta else "",
            actor_type=("human" if meta and int(meta.actor_type) == 1 else "service" if meta and int(meta.actor_type) == 2 else ""),
            capability=meta.capability if meta else "",
            risk_tags=risk_tags,
            requires=list(meta.requires) if meta else [],
            pack_id=meta.pack_id if meta else "",
            secrets_present=secrets_present(risk_tags, dict(req.labels)),
            mcp=extract_mcp_request(dict(req.labels)),
        )

    def evaluate(self, req: PolicyCheckRequest) -> PolicyCheckResponse:
        with self._mu:
            policy = self._policy
            snapshot = self._snapshot

        cache_key = None
        if self._cache_ttl > 0:
            cache_key = self._cache_key(req, snapshot)
            cached = self._cache_get(cache_key)
            if cached is not None:
                out = PolicyCheckResponse.decode(cached.encode())
                if out.approval_required:
                    out.approval_ref = req.job_id
                out.policy_snapshot = snapshot
                return out

        topic = (req.topic or "").strip()
        if not topic:
            return PolicyCheckResponse(decision=DecisionType.DENY, reason="missing topic")
        if not topic.startswith("job."):
            return PolicyCheckResponse(decision=DecisionType.DENY, reason="unsupported topic")

        inp = self.input_from_request(req)

        pd = pol.PolicyDecision(decision=pol.DECISION_ALLOW)
        if policy is not None:
            pd = policy.evaluate(inp)
        resp = self.finish_response(pd, inp, topic, req, policy, snapshot)
        if cache_key:
            cached = PolicyCheckResponse.decode(resp.encode())
            cached.approval_ref = ""
            self._cache_put(cache_key, cached)
        return resp

    def finish_response(
        self,
        pd: pol.PolicyDecision,
        inp: pol.PolicyInput,
        topic: str,
        req: PolicyCheckRequest,
        policy: Optional[pol.SafetyPolicy],
        snapshot: str,
    ) -> PolicyCheckResponse:
        """The post-first-match half of evaluate(): tenant MCP gate, decision
        normalization, effective-config restrictions, response assembly.
        Shared verbatim with the batched device path (runtime/device_gate.py)
        so kernel==device decisions are the same code for everything after
        the rule match (which is what the K1 kernel computes)."""
        if policy is not None:
            tp = policy.tenants.get(inp.tenant)
            if tp is not None:
                ok, mcp_reason = pol.mcp_allowed(tp.mcp, inp.mcp)
                if not ok:
                    pd.decision = pol.DECISION_DENY
                    pd.reason = mcp_reason

        decision = DecisionType.ALLOW
        reason = ""
        constraints = pd.constraints
        if pd.decision == pol.DECISION_DENY:
            decision, reason = DecisionType.DENY, pd.reason
        elif pd.decision == pol.DECISION_REQUIRE_APPROVAL:
            decision, reason = DecisionType.REQUIRE_HUMAN, pd.reason
        elif pd.decision == pol.DECISION_THROTTLE:
            decision, reason = DecisionType.THROTTLE, pd.reason
        elif pd.decision == pol.DECISION_ALLOW_WITH_CONSTRAINTS:
            decision = DecisionType.ALLOW_WITH_CONSTRAINTS
        elif pd.decision == pol.DECISION_ALLOW and constraints is not None:
            decision = DecisionType.ALLOW_WITH_CONSTRAINTS

        # effective-config restrictions (kernel.go:218-231)
        eff = parse_effective_safety(req.effective_config)
        if eff is not None:
            denied = eff.get("denied_topics", []) or []
            allowed = eff.get("allowed_topics", []) or []
            if any(pol.match_topic(p, topic) for p in denied):
                decision, reason = DecisionType.DENY, f"topic '{topic}' denied by effective config"
            if allowed and not any(pol.match_topic(p, topic) for p in allowed):
                decision, reason = DecisionType.DENY, f"topic '{topic}' not allowed by effective config"
            eff_mcp = pol.MCPPolicy.from_dict(eff.get("mcp"))
            ok, mcp_reason = pol.mcp_allowed(eff_mcp, inp.mcp)
            if not ok:
                decision, reason = DecisionType.DENY, mcp_reason

        approval_required = pd.approval_required or decision == DecisionType.REQUIRE_HUMAN
        return PolicyCheckResponse(
            decision=decision,
            reason=reason,
            policy_snapshot=snapshot,
            rule_id=pd.rule_id,
            constraints=constraints,
            approval_required=approval_required,
            approval_ref=req.job_id if approval_required else "",
            remediations=list(pd.remediations),
        )

    # -- decision cache (kernel.go:259-303) --------------------------------------
    def _cache_key(self, req: PolicyCheckRequest, snapshot: str) -> str:
        clone = PolicyCheckRequest.decode(req.encode())
        clone.job_id = ""
        return snapshot + ":" + hashlib.sha256(clone.encode()).hexdigest()

    def _cache_get(self, key: str) -> Optional[PolicyCheckResponse]:
        now = self._clock.now()
        with self._mu:
            e = self._cache.get(key)
            if e is None:
                return None
            if now > e.expires:
                del self._cache[key]
                return None
            return e.resp

    def _cache_put(self, key: str, resp: PolicyCheckResponse) -> None:
        with self._mu:
            if len(self._cache) > 65536:
                self._cache.clear()
            self._cache[key] = _CacheEntry(resp, self._clock.now() + self._cache_ttl)

    def cache_size(self) -> int:
        with self._mu:
            return len(self._cache)


def _clone_policy(p: Optional[pol.SafetyPolicy]) -> pol.SafetyPolicy:
    if p is None:
        return pol.SafetyPolicy()
    return pol.SafetyPolicy(
        version=p.version,
        rules=list(p.rules),
        default_tenant=p.default_tenant,
        tenants=dict(p.tenants),
    )


def _policy_hash(p: Optional[pol.SafetyPolicy]) -> str:
    if p is None:
        return "0"
    h = hashlib.sha256()
    for r in p.rules:
        h.update(r.id.encode())
        h.update(r.decision.encode())
        for t in r.match.topics:
            h.update(t.encode())
        for t in r.match.tenants:
            h.update(t.encode())
    for t in sorted(p.tenants):
        h.update(t.encode())
    return h.hexdigest()[:16]


class AllowAllSafety:
    """Allow-all checker: the reference test seam NewSafetyBasic
    (scheduler/safety_basic_test.go) used for CPU-spine and unit tests."""

    snapshot = "allow-all"

    def check(self, req: PolicyCheckRequest) -> PolicyCheckResponse:
        return PolicyCheckResponse(decision=DecisionType.ALLOW, policy_snapshot=self.snapshot)

    evaluate = check
    simulate = check
    explain = check

    def list_snapshots(self):
        return [self.snapshot]
