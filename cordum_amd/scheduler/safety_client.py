"""Scheduler-side safety checker: in-process kernel call + circuit breaker.

Oracle: scheduler/safety_client.go:18-208 — builds PolicyCheckRequest from the
JobRequest (tenant via ExtractTenant; CORDUM_EFFECTIVE_CONFIG env forwarded as
effective_config bytes), 2s timeout with deny-on-error, circuit breaker
(3 failures -> open 30s -> half-open 3 probes -> close after 2 successes).

In the MI355X build the kernel is fused into the same process (SURVEY.md §2.1
#7), so the gRPC hop is a function call; the breaker is kept because the
checker interface admits out-of-process implementations (and the breaker
behavior is part of the tested contract, safety_client_test.go).
"""
from __future__ import annotations

import threading

from ..protocol.capv2 import (
    DecisionType,
    JobRequest,
    PolicyCheckRequest,
    PolicyCheckResponse,
)
from ..store.job_store import SafetyDecisionRecord
from ..utils.clock import Clock, SYSTEM_CLOCK

DEFAULT_TENANT = "default"

SAFETY_CIRCUIT_FAIL_BUDGET = 3
SAFETY_CIRCUIT_OPEN_FOR_S = 30.0
SAFETY_CIRCUIT_HALF_OPEN_MAX = 3
SAFETY_CIRCUIT_CLOSE_AFTER = 2

_CLOSED, _OPEN, _HALF_OPEN = 0, 1, 2

DECISION_NAMES = {
    DecisionType.ALLOW: "allow",
    DecisionType.DENY: "deny",
    DecisionType.REQUIRE_HUMAN: "require_approval",
    DecisionType.THROTTLE: "throttle",
    DecisionType.ALLOW_WITH_CONSTRAINTS: "allow_with_constraints",
}


def extract_tenant(req: JobRequest) -> str:
    if req.tenant_id:
        return req.tenant_id
    if req.env and req.env.get("tenant_id"):
        return req.env["tenant_id"]
    return DEFAULT_TENANT


def build_check_request(req: JobRequest) -> PolicyCheckRequest:
    check = PolicyCheckRequest(
        job_id=req.job_id,
        topic=req.topic,
        tenant=extract_tenant(req),
        principal_id=req.principal_id,
        priority=req.priority,
        budget=req.budget,
        labels=dict(req.labels),
        memory_id=req.memory_id,
        meta=req.meta,
    )
    eff = (req.env or {}).get("CORDUM_EFFECTIVE_CONFIG", "")
    if eff:
        check.effective_config = eff.encode("utf-8")
    return check


def record_from_response(resp: PolicyCheckResponse) -> SafetyDecisionRecord:
    """PolicyCheckResponse -> stored decision record (shared with the batched
    device gate so both paths persist identical records)."""
    return SafetyDecisionRecord(
        decision=DECISION_NAMES.get(resp.decision, ""),
        reason=resp.reason,
        rule_id=resp.rule_id,
        policy_snapshot=resp.policy_snapshot,
        constraints=resp.constraints,
        approval_required=resp.approval_required,
        approval_ref=resp.approval_ref,
        remediations=list(resp.remediations),
    )


class SafetyChecker:
    """Wraps a kernel-like object exposing check(PolicyCheckRequest)."""

    def __init__(self, kernel, clock: Clock = SYSTEM_CLOCK):
        self._kernel = kernel
        self._clock = clock
        self._mu = threading.Lock()
        self._state = _CLOSED
        self._failures = 0
        self._successes = 0
        self._open_until = 0.0
        self._half_open_allowed = 0

    # -- breaker -------------------------------------------------------------
    def _is_open(self) -> bool:
        with self._mu:
            if self._state != _OPEN:
                return False
            if self._clock.now() >= self._open_until:
                self._state = _HALF_OPEN
                self._half_open_allowed = SAFETY_CIRCUIT_HALF_OPEN_MAX
                self._successes = 0
                return False
            return True

    def _allow_half_open(self) -> bool:
        with self._mu:
            if self._state != _HALF_OPEN:
                return True
            if self._half_open_allowed <= 0:
                return False
            self._half_open_allowed -= 1
            return True

    def _record_failure(self) -> None:
        with self._mu:
            self._failures += 1
            self._successes = 0
            if self._state == _HALF_OPEN or self._failures >= SAFETY_CIRCUIT_FAIL_BUDGET:
                self._state = _OPEN
                self._open_until = self._clock.now() + SAFETY_CIRCUIT_OPEN_FOR_S
                self._failures = 0

    def _record_success(self) -> None:
        with self._mu:
            self._failures = 0
            if self._state == _HALF_OPEN:
                self._successes += 1
                if self._successes >= SAFETY_CIRCUIT_CLOSE_AFTER:
                    self._state = _CLOSED
                    self._successes = 0
            else:
                self._state = _CLOSED

    # -- check ---------------------------------------------------------------
    def check(self, req: JobRequest) -> SafetyDecisionRecord:
        if self._is_open():
            return SafetyDecisionRecord(decision="deny", reason="safety kernel circuit open")
        if not self._allow_half_open():
            return SafetyDecisionRecord(decision="deny", reason="safety kernel circuit half-open (throttled)")
        try:
            resp: PolicyCheckResponse = self._kernel.check(build_check_request(req))
        except Exception as e:  # deny-on-error
            self._record_failure()
            return SafetyDecisionRecord(decision="deny", reason=f"safety kernel error: {e}")
        self._record_success()
        return record_from_response(resp)

    def list_snapshots(self):
        return self._kernel.list_snapshots()
