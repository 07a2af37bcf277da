"""Scheduler engine: safety gate → routing → dispatch → state machine → DLQ.

Semantics oracle: core/controlplane/scheduler/engine.go —
 Start :109 (subscriptions: heartbeats broadcast, submit/result/cancel in
 queue group "cordum-scheduler"), HandlePacket :127, handleJobRequest :203
 (per-job lock, early-exit on terminal/DISPATCHED/RUNNING, trace/meta/request
 persistence, PENDING), processJob :277 (effective config attach → safety
 gate → constraints → max-retries → tenant concurrency → deadline →
 PickSubject → SCHEDULED → publish → DISPATCHED → RUNNING),
 checkSafetyDecision :474-547 (approval label honored only when the stored
 decision required approval AND the job hash matches), handleJobResult
 :549-617 (terminal dedup, result ptr, DLQ on non-success), applyConstraints
 :674-706, emitDLQ :798-819, publishCancel :779-796.

The per-tick batched path (policy kernel + scoring kernel over the device
job table) lives in ops/pipeline.py; it implements the same decision logic
for homogeneous batches and falls back to this engine for anything that
needs per-job host logic (approvals, constraints injection with effective
config, remediation).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional

from ..bus import Bus, RetryAfter
from ..protocol import JobState, is_terminal
from ..protocol import subjects as subj
from ..protocol.capv2 import (
    BusPacket,
    JobCancel,
    JobRequest,
    JobResult,
    JobStatus,
    PolicyConstraints,
)
from ..store import InvalidTransition, JobStore, SafetyDecisionRecord
from ..store.configsvc import ConfigService
from ..utils.canonical_json import canonical_json
from ..utils.clock import Clock, SYSTEM_CLOCK
from ..utils.hashing import job_hash
from . import errors as errs
from .registry import WorkerRegistry
from .strategy import Strategy

QUEUE_GROUP = "cordum-scheduler"
JOB_LOCK_TTL_S = 30.0
RETRY_DELAY_STORE_S = 1.0
RETRY_DELAY_PUBLISH_S = 2.0
RETRY_DELAY_NO_WORKERS_S = 2.0
SAFETY_THROTTLE_DELAY_S = 5.0
SENDER_ID = "cordum-scheduler"


@dataclass
class Metrics:
    jobs_received: Dict[str, int] = field(default_factory=dict)
    jobs_dispatched: Dict[str, int] = field(default_factory=dict)
    jobs_completed: Dict[tuple, int] = field(default_factory=dict)
    safety_denied: Dict[str, int] = field(default_factory=dict)

    def inc_received(self, topic):
        self.jobs_received[topic] = self.jobs_received.get(topic, 0) + 1

    def inc_dispatched(self, topic):
        self.jobs_dispatched[topic] = self.jobs_dispatched.get(topic, 0) + 1

    def inc_completed(self, topic, status):
        self.jobs_completed[(topic, status)] = self.jobs_completed.get((topic, status), 0) + 1

    def inc_safety_denied(self, topic):
        self.safety_denied[topic] = self.safety_denied.get(topic, 0) + 1


class Engine:
    def __init__(
        self,
        bus: Bus,
        job_store: JobStore,
        safety,  # SafetyChecker-like: .check(JobRequest) -> SafetyDecisionRecord
        registry: WorkerRegistry,
        strategy: Strategy,
        configsvc: Optional[ConfigService] = None,
        metrics: Optional[Metrics] = None,
        clock: Clock = SYSTEM_CLOCK,
    ):
        self.bus = bus
        self.job_store = job_store
        self.safety = safety
        self.registry = registry
        self.strategy = strategy
        self.configsvc = configsvc
        self.metrics = metrics or Metrics()
        self.clock = clock
        self._stopped = False
        self._subs = []

    # -- wiring ---------------------------------------------------------------
    def start(self) -> None:
        """engine.go:109-125: heartbeats broadcast (no group); submit/result/
        cancel in the scheduler queue group."""
        self._subs = [
            self.bus.subscribe(subj.SUBJECT_HEARTBEAT, self.handle_packet),
            self.bus.subscribe(subj.SUBJECT_SUBMIT, self.handle_packet, queue_group=QUEUE_GROUP),
            self.bus.subscribe(subj.SUBJECT_RESULT, self.handle_packet, queue_group=QUEUE_GROUP),
            self.bus.subscribe(subj.SUBJECT_CANCEL, self.handle_packet, queue_group=QUEUE_GROUP),
        ]

    def stop(self) -> None:
        self._stopped = True
        for s in self._subs:
            s.unsubscribe()

    # -- packet dispatch -------------------------------------------------------
    def handle_packet(self, subject: str, pkt: BusPacket) -> None:
        if pkt.heartbeat is not None:
            self.registry.update(pkt.heartbeat)
            return
        if pkt.job_request is not None and subject == subj.SUBJECT_SUBMIT:
            self.handle_job_request(pkt.job_request, pkt.trace_id)
            return
        if pkt.job_result is not None:
            self.handle_job_result(pkt.job_result)
            return
        if pkt.job_cancel is not None:
            # bus-delivered cancel: store transition only — republishing here
            # would echo the broadcast back onto the same subject
            self.job_store.cancel_job(pkt.job_cancel.job_id)

    # -- submit path -----------------------------------------------------------
    def handle_job_request(self, req: JobRequest, trace_id: str = "") -> None:
        job_id = (req.job_id or "").strip()
        topic = (req.topic or "").strip()
        if not job_id or not topic:
            if job_id:
                self._set_state_quiet(job_id, JobState.FAILED)
            self.metrics.inc_completed(topic, "FAILED")
            return

        with self.job_store.job_lock(job_id, owner=SENDER_ID) as locked:
            if not locked or self._stopped:
                return
            current = self.job_store.get_state(job_id)
            if is_terminal(current) or current in (JobState.DISPATCHED, JobState.RUNNING):
                return

            self.metrics.inc_received(topic)
            if trace_id:
                self.job_store.add_job_to_trace(trace_id, job_id)
            self.job_store.set_job_meta(
                job_id,
                topic=topic,
                tenant=_tenant_of(req),
                trace_id=trace_id,
                memory_id=req.memory_id,
                principal=req.principal_id,
                priority=int(req.priority),
                labels=dict(req.labels),
            )
            self.job_store.set_job_request(job_id, req)
            if current == JobState.UNSPECIFIED:
                self.job_store.set_state(job_id, JobState.PENDING)
            self.process_job(req, trace_id)

    def process_job(self, req: JobRequest, trace_id: str = "") -> None:
        job_id = (req.job_id or "").strip()
        topic = (req.topic or "").strip()

        self.attach_effective_config(req)
        record = self.check_safety_decision(req)
        if not self.apply_decision(req, record):
            return
        if not self.pre_dispatch_checks(req, record):
            return

        # routing (engine.go:392-407)
        workers = self.registry.snapshot()
        try:
            subject = self.strategy.pick_subject(req, workers)
        except Exception as e:
            if errs.is_retryable(e):
                raise RetryAfter(RETRY_DELAY_NO_WORKERS_S, str(e))
            self._set_state_quiet(job_id, JobState.FAILED)
            self.metrics.inc_completed(topic, "FAILED")
            self.emit_dlq(job_id, topic, JobStatus.FAILED, str(e), errs.reason_code_for(e))
            return

        self.dispatch(req, trace_id, subject)

    def apply_decision(self, req: JobRequest, record: SafetyDecisionRecord) -> bool:
        """Decision branches (engine.go:298-347): returns True to continue to
        dispatch, False when the job reached a decision-terminal state; raises
        RetryAfter for throttle."""
        job_id = (req.job_id or "").strip()
        topic = (req.topic or "").strip()
        decision = record.decision
        if decision in ("allow", "allow_with_constraints"):
            if record.constraints is not None:
                apply_constraints(req, record.constraints)
            return True
        if decision == "throttle":
            raise RetryAfter(SAFETY_THROTTLE_DELAY_S, f"safety throttle: {record.reason}")
        if decision == "require_approval":
            self._set_state_quiet(job_id, JobState.APPROVAL_REQUIRED)
            return False
        if decision == "deny":
            self._set_state_quiet(job_id, JobState.DENIED)
            self.metrics.inc_safety_denied(topic)
            self.emit_dlq(job_id, topic, JobStatus.DENIED, record.reason, errs.REASON_SAFETY_DENIED)
            return False
        self._set_state_quiet(job_id, JobState.DENIED)
        self.metrics.inc_safety_denied(topic)
        self.emit_dlq(job_id, topic, JobStatus.DENIED, record.reason, errs.REASON_SAFETY_UNKNOWN)
        return False

    def pre_dispatch_checks(self, req: JobRequest, record: SafetyDecisionRecord) -> bool:
        """Max retries + tenant concurrency + deadline registration
        (engine.go:349-390); raises RetryAfter on tenant limit."""
        job_id = (req.job_id or "").strip()
        topic = (req.topic or "").strip()
        max_retries = _max_retries(record.constraints)
        if max_retries > 0:
            attempts = int(self.job_store.get_job_meta(job_id).get("attempts", 0) or 0)
            if attempts >= max_retries + 1:
                reason = f"max retries exceeded (attempts={attempts}, max_retries={max_retries})"
                self._set_state_quiet(job_id, JobState.FAILED)
                self.emit_dlq(job_id, topic, JobStatus.FAILED, reason, errs.REASON_MAX_RETRIES)
                return False

        max_concurrent = _max_concurrent(record.constraints)
        if max_concurrent > 0:
            tenant = _tenant_of(req)
            if self.job_store.tenant_active_count(tenant) > max_concurrent:
                # the job itself is PENDING and counted; limit is on others + self
                raise RetryAfter(RETRY_DELAY_NO_WORKERS_S, "tenant limit")

        if req.budget is not None and req.budget.deadline_ms > 0:
            at = self.clock.now_micros() + req.budget.deadline_ms * 1000
            self.job_store.set_deadline(job_id, at)
        return True

    def dispatch(self, req: JobRequest, trace_id: str, subject: str) -> None:
        """SCHEDULED -> publish -> DISPATCHED -> RUNNING (engine.go:409-441)."""
        job_id = (req.job_id or "").strip()
        self.job_store.set_state(job_id, JobState.SCHEDULED)
        packet = BusPacket(trace_id=trace_id, protocol_version=1, job_request=req)
        self.bus.publish(subject, packet)
        self.metrics.inc_dispatched((req.topic or "").strip())
        self.job_store.set_state(job_id, JobState.DISPATCHED)
        self.job_store.set_state(job_id, JobState.RUNNING)

    # -- safety ---------------------------------------------------------------
    def approval_resume_record(self, req: JobRequest) -> Optional[SafetyDecisionRecord]:
        """engine.go:484-522: the approval label is honored only when the
        stored decision required approval AND the job hash matches. Returns
        the allow record, or None (proceed to a fresh safety check)."""
        job_id = (req.job_id or "").strip()
        approved = (req.labels or {}).get("approval_granted", "").strip().lower() == "true"
        if not approved:
            return None
        prev = self.job_store.get_safety_decision(job_id)
        if prev is not None and (prev.approval_required or prev.decision == "require_approval") and prev.job_hash:
            h = job_hash(req)
            if h == prev.job_hash:
                record = SafetyDecisionRecord(
                    decision="allow",
                    reason="approval granted",
                    checked_at=self.clock.now_micros(),
                    constraints=prev.constraints,
                    policy_snapshot=prev.policy_snapshot,
                    rule_id=prev.rule_id,
                    job_hash=prev.job_hash,
                )
                self.job_store.set_safety_decision(job_id, record)
                return record
            # hash mismatch: approval label ignored
        return None

    def finalize_safety_record(self, req: JobRequest, record: SafetyDecisionRecord) -> SafetyDecisionRecord:
        """Approval-binding fixups + persistence, shared by the per-job host
        check and the batched device gate (runtime/device_dispatch.py)."""
        job_id = (req.job_id or "").strip()
        if not record.checked_at:
            record.checked_at = self.clock.now_micros()
        if record.approval_required and record.decision in ("allow", "allow_with_constraints"):
            record.decision = "require_approval"
        if record.decision == "require_approval" or record.approval_required:
            record.job_hash = job_hash(req)
        self.job_store.set_safety_decision(job_id, record)
        return record

    def check_safety_decision(self, req: JobRequest) -> SafetyDecisionRecord:
        resumed = self.approval_resume_record(req)
        if resumed is not None:
            return resumed
        return self.finalize_safety_record(req, self.safety.check(req))

    # -- results ---------------------------------------------------------------
    def handle_job_result(self, res: JobResult) -> None:
        job_id = (res.job_id or "").strip()
        if not job_id:
            return
        with self.job_store.job_lock(job_id, owner=SENDER_ID) as locked:
            if not locked:
                return
            topic = self.job_store.get_job_meta(job_id).get("topic", "") or "unknown"
            state_now = self.job_store.get_state(job_id)
            if is_terminal(state_now):
                return  # duplicate result
            status = res.status
            state = {
                JobStatus.SUCCEEDED: JobState.SUCCEEDED,
                JobStatus.FAILED: JobState.FAILED,
                JobStatus.TIMEOUT: JobState.TIMEOUT,
                JobStatus.DENIED: JobState.DENIED,
                JobStatus.CANCELLED: JobState.CANCELLED,
            }.get(status, JobState.FAILED)
            try:
                self.job_store.set_state(job_id, state)
            except InvalidTransition:
                raise RetryAfter(RETRY_DELAY_STORE_S, "state transition race")
            if res.result_ptr:
                self.job_store.set_result_ptr(job_id, res.result_ptr)
            if res.worker_id:
                self.job_store.set_worker(job_id, res.worker_id)
            if res.error_code or res.error_message:
                self.job_store.set_error(job_id, res.error_code, res.error_message)
            self.metrics.inc_completed(topic, state.name)
            if state != JobState.SUCCEEDED:
                self.emit_dlq(job_id, topic, status, res.error_message, res.error_code)

    # -- cancel ----------------------------------------------------------------
    def cancel_job(self, job_id: str, reason: str = "cancelled by request") -> bool:
        ok = self.job_store.cancel_job(job_id)
        self.publish_cancel(job_id, reason)
        return ok

    def publish_cancel(self, job_id: str, reason: str) -> None:
        pkt = BusPacket(trace_id=job_id, protocol_version=1, job_cancel=JobCancel(job_id=job_id, reason=reason))
        self.bus.publish(subj.SUBJECT_CANCEL, pkt)

    # -- DLQ -------------------------------------------------------------------
    def emit_dlq(self, job_id: str, topic: str, status: JobStatus, reason: str, reason_code: str) -> None:
        if not job_id:
            return
        pkt = BusPacket(
            trace_id=job_id,
            protocol_version=1,
            job_result=JobResult(job_id=job_id, status=status, error_code=reason_code, error_message=reason or ""),
        )
        self.bus.publish(subj.SUBJECT_DLQ, pkt)

    # -- effective config -------------------------------------------------------
    def attach_effective_config(self, req: JobRequest) -> None:
        if self.configsvc is None:
            return
        env = req.env or {}
        snap = self.configsvc.effective(
            org=_tenant_of(req),
            team=env.get("team_id", ""),
            workflow=req.workflow_id,
            step=env.get("step_id", ""),
        )
        if not snap.config:
            return
        if req.env is None:
            req.env = {}
        req.env["CORDUM_EFFECTIVE_CONFIG"] = canonical_json(snap.config)

    # -- helpers -----------------------------------------------------------------
    def _set_state_quiet(self, job_id: str, state: JobState) -> None:
        try:
            self.job_store.set_state(job_id, state)
        except (InvalidTransition, ValueError):
            pass


def _tenant_of(req: JobRequest) -> str:
    from .safety_client import extract_tenant

    return extract_tenant(req)


def _max_retries(constraints: Optional[PolicyConstraints]) -> int:
    if constraints is None or constraints.budgets is None:
        return 0
    return int(constraints.budgets.max_retries)


def _max_concurrent(constraints: Optional[PolicyConstraints]) -> int:
    if constraints is None or constraints.budgets is None:
        return 0
    return int(constraints.budgets.max_concurrent_jobs)


def apply_constraints(req: JobRequest, constraints: PolicyConstraints) -> None:
    """engine.go:674-706: inject constraint env vars + clamp Budget.DeadlineMs."""
    import json

    if req.env is None:
        req.env = {}
    req.env["CORDUM_POLICY_CONSTRAINTS"] = json.dumps(constraints.to_dict(), sort_keys=True, separators=(",", ":"))
    if constraints.redaction_level:
        req.env["CORDUM_REDACTION_LEVEL"] = constraints.redaction_level
    budgets = constraints.budgets
    if budgets is not None:
        from ..protocol.capv2 import Budget

        if req.budget is None:
            req.budget = Budget()
        if budgets.max_runtime_ms > 0:
            if req.budget.deadline_ms == 0 or req.budget.deadline_ms > budgets.max_runtime_ms:
                req.budget.deadline_ms = budgets.max_runtime_ms
        if budgets.max_artifact_bytes > 0:
            req.env["CORDUM_MAX_ARTIFACT_BYTES"] = str(budgets.max_artifact_bytes)
        if budgets.max_concurrent_jobs > 0:
            req.env["CORDUM_MAX_CONCURRENT_JOBS"] = str(budgets.max_concurrent_jobs)
        if budgets.max_retries > 0:
            req.env["CORDUM_MAX_RETRIES"] = str(budgets.max_retries)
