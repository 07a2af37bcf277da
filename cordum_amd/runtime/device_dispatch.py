"""Device dispatch engine: the scheduler's hot path on GPU kernels.

This is the seam that fuses the serving path onto the device data plane
(the round-1 gap): jobs submitted through the gateway land here via
`sys.job.submit` exactly like the host engine, but the per-job safety check
and the per-job worker-scoring loop — the two hot loops the reference runs
in Go (kernel.go:129-257, strategy_least_loaded.go:92-119) — are batched
through the K1 policy kernel and the K2 least-loaded kernel:

  handle_job_request   host: lock, meta/trace/request persistence, PENDING
                       (engine.go:203-275) — then BUFFER instead of inline
  flush()              1. approval-resume short-circuits (host, per job)
                       2. K1 batched first-match -> PolicyCheckResponses
                          (runtime/device_gate.py; one kernel launch + one
                          D2H read for the whole batch)
                       3. decision semantics per job (engine.go:298-390):
                          deny->DLQ, approval hold, throttle->delayed requeue
                       4. K2 batched least-loaded pick over the packed worker
                          table (ops/worker_table.py; one launch + one D2H)
                       5. dispatch: SCHEDULED -> worker subject -> DISPATCHED
                          -> RUNNING; device-pool workers execute the batch
                          with the echo kernel (ops/worker_pool.py)

Retry semantics: the host engine raises RetryAfter inside the bus delivery
and the bus NAK-redelivers (bus/nats.go:146-168). flush() runs outside any
delivery, so retryable outcomes (throttle, tenant limit, no workers,
overload) are requeued with bus.publish_after under the same max-deliver
budget the bus enforces (DEFAULT_MAX_DELIVER) — at-least-once with bounded
redelivery, identical to the JetStream contract. Jobs that exhaust the
budget stay PENDING and are re-driven by the pending replayer, as in the
reference.

Host fallbacks (correctness first, device speed second): a policy whose
vocabulary overflows the compiled bitset words, a worker fleet with >63
pools/label pairs, or a non-least-loaded strategy fall back to the host code
paths per job. Both paths share the same decision/record/dispatch code, so
the fallback cannot change semantics.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional, Tuple

from ..bus import DEFAULT_MAX_DELIVER, RetryAfter
from ..protocol import JobState, is_terminal
from ..protocol import subjects as subj
from ..protocol.capv2 import BusPacket, JobRequest, JobStatus
from ..scheduler import errors as errs
from ..scheduler.engine import (
    RETRY_DELAY_NO_WORKERS_S,
    SENDER_ID,
    Engine,
)
from ..scheduler.safety_client import build_check_request, record_from_response
from ..scheduler.strategy import is_overloaded, matches_labels
from ..protocol.subjects import worker_subject


class DeviceDispatchEngine(Engine):
    def __init__(self, *args, gate=None, worker_table=None, **kw):
        super().__init__(*args, **kw)
        self.gate = gate  # DeviceBatchGate
        self.worker_table = worker_table  # DeviceWorkerTable
        self._flush_mu = threading.Lock()
        self._pending: List[Tuple[JobRequest, str]] = []
        self._redeliveries: Dict[str, int] = {}
        self.flushes = 0
        self.device_routed = 0
        self.host_routed = 0

    # -- submit path: buffer instead of inline process_job ------------------------
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
                tenant=_tenant(req),
                trace_id=trace_id,
                memory_id=req.memory_id,
                principal=req.principal_id,
                priority=int(req.priority),
                labels=dict(req.labels),
            )
            self.job_store.set_job_request(job_id, req)
            if current == JobState.UNSPECIFIED:
                self.job_store.set_state(job_id, JobState.PENDING)
        with self._flush_mu:
            self._pending.append((req, trace_id))

    def pending_count(self) -> int:
        with self._flush_mu:
            return len(self._pending)

    # -- requeue (NAK analog) -------------------------------------------------------
    def _requeue(self, req: JobRequest, trace_id: str, delay_s: float, cause: str) -> None:
        job_id = req.job_id
        n = self._redeliveries.get(job_id, 0) + 1
        if n >= DEFAULT_MAX_DELIVER:
            # delivery budget exhausted: job stays PENDING; the pending
            # replayer re-drives it later (pending_replayer.go)
            self._redeliveries.pop(job_id, None)
            return
        self._redeliveries[job_id] = n
        pkt = BusPacket(trace_id=trace_id, protocol_version=1, job_request=req)
        self.bus.publish_after(delay_s, subj.SUBJECT_SUBMIT, pkt)

    def _done(self, job_id: str) -> None:
        self._redeliveries.pop(job_id, None)

    # -- the batched flush -------------------------------------------------------------
    def flush(self) -> int:
        """Process every buffered submission through the batched device path.
        Returns the number of jobs processed (0 = nothing buffered)."""
        with self._flush_mu:
            batch, self._pending = self._pending, []
        if not batch:
            return 0
        self.flushes += 1

        # 1) effective config + approval-resume short-circuit (host per job)
        records: List[Optional[object]] = [None] * len(batch)
        eval_idx: List[int] = []
        for i, (req, _tr) in enumerate(batch):
            self.attach_effective_config(req)
            rec = self.approval_resume_record(req)
            if rec is not None:
                records[i] = rec
            else:
                eval_idx.append(i)

        # 2) batched K1 safety gate
        if eval_idx:
            if self.gate is not None:
                resps = self.gate.evaluate_batch(
                    [build_check_request(batch[i][0]) for i in eval_idx]
                )
                for i, resp in zip(eval_idx, resps):
                    records[i] = self.finalize_safety_record(
                        batch[i][0], record_from_response(resp)
                    )
            else:
                for i in eval_idx:
                    records[i] = self.check_safety_decision(batch[i][0])

        # 3) decision semantics + pre-dispatch checks (host per job)
        routable: List[int] = []
        for i, (req, trace_id) in enumerate(batch):
            # a job can reach a terminal state between buffering and flush
            # (cancel, a concurrent duplicate) — the buffered copy is then
            # stale and must be dropped, like the host engine's early exit
            # (engine.go:226-237)
            if is_terminal(self.job_store.get_state(req.job_id)):
                self._done(req.job_id)
                continue
            rec = records[i]
            try:
                if not self.apply_decision(req, rec):
                    self._done(req.job_id)
                    continue
                if not self.pre_dispatch_checks(req, rec):
                    self._done(req.job_id)
                    continue
            except RetryAfter as ra:
                self._requeue(req, trace_id, ra.delay_s, ra.cause)
                continue
            routable.append(i)

        # 4) routing: K2 over the packed worker table, host fallbacks as needed
        if routable:
            self._route_and_dispatch(batch, routable)
        return len(batch)

    def _route_and_dispatch(self, batch, routable: List[int]) -> None:
        workers = self.registry.snapshot()
        table = self.worker_table
        resolve = getattr(self.strategy, "resolve", None)
        exact = bool(table is not None and resolve is not None) and table.pack(workers)

        device_jobs: List[int] = []  # indexes into batch
        pool_masks: List[int] = []
        label_masks: List[int] = []
        pools_of: Dict[int, List[str]] = {}

        for i in routable:
            req, trace_id = batch[i]
            if not exact:
                self._host_route(req, trace_id)
                continue
            try:
                eligible, required_labels, preferred = resolve(req)
            except Exception as e:
                self._route_error(req, trace_id, e)
                continue
            if preferred:
                hb = workers.get(preferred)
                if (
                    hb is not None
                    and hb.pool in set(eligible)
                    and matches_labels(hb, required_labels)
                    and not is_overloaded(hb)
                ):
                    self.dispatch(req, trace_id, worker_subject(preferred))
                    self._done(req.job_id)
                    self.host_routed += 1
                    continue
                # unhealthy/ineligible preferred worker falls through to scoring
            device_jobs.append(i)
            pools_of[i] = eligible
            pool_masks.append(table.pool_mask(eligible))
            label_masks.append(table.label_mask(required_labels))

        if not device_jobs:
            return
        picks = table.pick(pool_masks, label_masks).cpu().tolist()  # one D2H
        for i, pick in zip(device_jobs, picks):
            req, trace_id = batch[i]
            pools_str = ",".join(pools_of[i])
            if pick == -2:
                self._route_error(req, trace_id, errs.PoolOverloaded(f"pool {pools_str!r}"))
            elif pick < 0:
                self._route_error(req, trace_id, errs.NoWorkers(f"pool {pools_str!r}"))
            else:
                _wid, subject = table.subject_for(pick)
                self._dispatch_safe(req, trace_id, subject)
                self._done(req.job_id)
                self.device_routed += 1

    def _dispatch_safe(self, req: JobRequest, trace_id: str, subject: str) -> None:
        """dispatch() with the cancel race closed: a transition refused by
        the legality table (the job went terminal under us) is a no-op, not
        a batch-aborting error."""
        from ..store.job_store import InvalidTransition

        try:
            self.dispatch(req, trace_id, subject)
        except InvalidTransition:
            pass

    def _host_route(self, req: JobRequest, trace_id: str) -> None:
        try:
            subject = self.strategy.pick_subject(req, self.registry.snapshot())
        except Exception as e:
            self._route_error(req, trace_id, e)
            return
        self._dispatch_safe(req, trace_id, subject)
        self._done(req.job_id)
        self.host_routed += 1

    def _route_error(self, req: JobRequest, trace_id: str, e: Exception) -> None:
        job_id = (req.job_id or "").strip()
        topic = (req.topic or "").strip()
        if errs.is_retryable(e):
            self._requeue(req, trace_id, RETRY_DELAY_NO_WORKERS_S, str(e))
            return
        self._set_state_quiet(job_id, JobState.FAILED)
        self.metrics.inc_completed(topic, "FAILED")
        self.emit_dlq(job_id, topic, JobStatus.FAILED, str(e), errs.reason_code_for(e))
        self._done(job_id)


def _tenant(req: JobRequest) -> str:
    from ..scheduler.safety_client import extract_tenant

    return extract_tenant(req)
