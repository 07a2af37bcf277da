"""CAP worker runtime: the in-process worker pool loop.

Oracle: sdk/runtime/worker.go:20-320 — queue-subscribe pool subjects and the
direct `worker.<id>.jobs` subject, per-job cancel tracking, MaxParallelJobs
semaphore, automatic JobResult fill (status from handler outcome; CANCELLED
when a cancel arrived first), heartbeat emission (10s in the reference;
tick-driven here), progress emission helper.

This host runtime drives Python handler functions (echo workers, examples).
The GPU-resident worker pool — an HBM job queue consumed by a device kernel —
is ops/worker_pool.py; it presents the same Heartbeat rows to the registry.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..bus import Bus
from ..protocol import subjects as subj
from ..protocol.capv2 import (
    BusPacket,
    Heartbeat,
    JobProgress,
    JobRequest,
    JobResult,
    JobStatus,
)
from ..store.memory_store import MemoryStore
from ..utils.clock import Clock, SYSTEM_CLOCK

HEARTBEAT_INTERVAL_S = 10.0

# handler(req: JobRequest, ctx_blob: bytes|None) -> bytes (result payload)
Handler = Callable[[JobRequest, Optional[bytes]], bytes]


class HandlerError(Exception):
    def __init__(self, message: str, code: str = "handler_error"):
        super().__init__(message)
        self.code = code


@dataclass
class Worker:
    bus: Bus
    memory: MemoryStore
    worker_id: str
    handler: Handler
    pool: str = "default"
    topics: List[str] = field(default_factory=lambda: ["job.default"])
    capabilities: List[str] = field(default_factory=list)
    labels: Dict[str, str] = field(default_factory=dict)
    max_parallel_jobs: int = 4
    region: str = ""
    type: str = "worker"
    clock: Clock = SYSTEM_CLOCK

    def __post_init__(self):
        self._mu = threading.Lock()
        self._active = 0
        self._cancelled: set = set()
        self._subs = []
        self.jobs_handled = 0

    # -- lifecycle -------------------------------------------------------------
    def start(self) -> None:
        # deferred=True: jobs are consumed from the queue on bus.pump(), like a
        # real worker draining its NATS subscription — results can never race
        # ahead of the scheduler's own dispatch bookkeeping.
        for topic in self.topics:
            self._subs.append(
                self.bus.subscribe(topic, self._on_job, queue_group=f"pool.{self.pool}", deferred=True)
            )
        self._subs.append(self.bus.subscribe(subj.worker_subject(self.worker_id), self._on_job, deferred=True))
        self._subs.append(self.bus.subscribe(subj.SUBJECT_CANCEL, self._on_cancel))
        self.send_heartbeat()

    def stop(self) -> None:
        for s in self._subs:
            s.unsubscribe()

    # -- heartbeat --------------------------------------------------------------
    def heartbeat(self) -> Heartbeat:
        with self._mu:
            active = self._active
        return Heartbeat(
            worker_id=self.worker_id,
            region=self.region,
            type=self.type,
            active_jobs=active,
            capabilities=list(self.capabilities),
            pool=self.pool,
            max_parallel_jobs=self.max_parallel_jobs,
            labels=dict(self.labels),
        )

    def send_heartbeat(self) -> None:
        self.bus.publish(subj.SUBJECT_HEARTBEAT, BusPacket(heartbeat=self.heartbeat()))

    # -- cancel -----------------------------------------------------------------
    def _on_cancel(self, subject: str, pkt: BusPacket) -> None:
        if pkt.job_cancel is not None and pkt.job_cancel.job_id:
            with self._mu:
                self._cancelled.add(pkt.job_cancel.job_id)

    # -- job handling -------------------------------------------------------------
    def _on_job(self, subject: str, pkt: BusPacket) -> None:
        req = pkt.job_request
        if req is None or not req.job_id:
            return
        with self._mu:
            if self._active >= self.max_parallel_jobs:
                # at capacity: NAK so another pool member (or a later pump) takes it
                from ..bus import RetryAfter

                raise RetryAfter(0.05, "worker at capacity")
            self._active += 1
        try:
            self._run_job(req, pkt.trace_id)
        finally:
            with self._mu:
                self._active -= 1

    def _run_job(self, req: JobRequest, trace_id: str) -> None:
        job_id = req.job_id
        start = self.clock.now()
        with self._mu:
            if job_id in self._cancelled:
                self._cancelled.discard(job_id)
                self._publish_result(req, trace_id, JobStatus.CANCELLED, "", "cancelled", "job cancelled", start)
                return
        ctx_blob = None
        if req.context_ptr:
            try:
                ctx_blob = self.memory.get_pointer(req.context_ptr)
            except ValueError:
                ctx_blob = None
        try:
            payload = self.handler(req, ctx_blob)
            result_ptr = ""
            if payload is not None:
                result_ptr = self.memory.put_result(job_id, payload)
            self.jobs_handled += 1
            self._publish_result(req, trace_id, JobStatus.SUCCEEDED, result_ptr, "", "", start)
        except HandlerError as he:
            self._publish_result(req, trace_id, JobStatus.FAILED, "", he.code, str(he), start)
        except Exception as e:  # noqa: BLE001 — worker must always report
            self._publish_result(req, trace_id, JobStatus.FAILED, "", "handler_error", str(e), start)

    def _publish_result(
        self,
        req: JobRequest,
        trace_id: str,
        status: JobStatus,
        result_ptr: str,
        error_code: str,
        error_message: str,
        start: float,
    ) -> None:
        res = JobResult(
            job_id=req.job_id,
            status=status,
            result_ptr=result_ptr,
            worker_id=self.worker_id,
            execution_ms=int((self.clock.now() - start) * 1000),
            error_code=error_code,
            error_message=error_message,
        )
        self.bus.publish(subj.SUBJECT_RESULT, BusPacket(trace_id=trace_id, job_result=res))

    def publish_progress(self, job_id: str, progress: float, message: str = "") -> None:
        self.bus.publish(
            subj.SUBJECT_PROGRESS,
            BusPacket(job_progress=JobProgress(job_id=job_id, worker_id=self.worker_id, progress=progress, message=message)),
        )


def echo_handler(req: JobRequest, ctx_blob: Optional[bytes]) -> bytes:
    """The examples' echo worker (examples/python-worker/worker.py:1-18)."""
    return ctx_blob if ctx_blob is not None else b"{}"
