"""Device worker pool: an HBM-staged echo worker pool executed by kernel.

The host Worker (runtime/worker.py) runs Python handlers per job; this pool
presents the same contract to the control plane — heartbeats into the
registry, jobs taken from `worker.<id>.jobs` subjects, JobResults published
on `sys.job.result` — but executes the whole batch on the GPU: context blobs
are staged into an int32 arena, one echo kernel launch copies payload and
computes per-job checksums in HBM, and the results are read back and
persisted as `res:<job_id>` blobs.

Echo is the reference's own benchmark workload (examples/python-worker/
worker.py, BASELINE configs #1/#2); heavier handlers stay host-side via
runtime/worker.py — the two pool kinds coexist on the same bus.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional, Tuple

import torch

from ..bus import Bus
from ..protocol import subjects as subj
from ..protocol.capv2 import BusPacket, Heartbeat, JobRequest, JobResult, JobStatus
from ..store.memory_store import MemoryStore
from ..utils.clock import Clock, SYSTEM_CLOCK


class DeviceWorkerPool:
    def __init__(
        self,
        bus: Bus,
        memory: MemoryStore,
        device: torch.device,
        ext,
        n_workers: int = 4,
        pool: str = "default",
        topics: Optional[List[str]] = None,
        rank: int = 0,
        max_parallel_jobs: int = 256,
        max_payload_bytes: int = 1 << 20,
        clock: Clock = SYSTEM_CLOCK,
    ):
        self.bus = bus
        self.memory = memory
        self.device = torch.device(device)
        self.ext = ext
        self.pool = pool
        self.topics = list(topics or ["job.default"])
        self.clock = clock
        self.max_parallel_jobs = max_parallel_jobs
        self.max_payload_bytes = max_payload_bytes
        self.worker_ids = [f"dev{rank}-w{i}" for i in range(n_workers)]
        self._widx = {w: i for i, w in enumerate(self.worker_ids)}
        self._mu = threading.Lock()
        self._staged: List[Tuple[JobRequest, str, int]] = []  # (req, trace_id, widx)
        self._cancelled: set = set()
        self._active = [0] * n_workers
        self._subs = []
        self.jobs_executed = 0
        self.batches_executed = 0

    # -- lifecycle ---------------------------------------------------------------
    def start(self) -> "DeviceWorkerPool":
        for i, wid in enumerate(self.worker_ids):
            self._subs.append(
                self.bus.subscribe(subj.worker_subject(wid), self._make_on_job(i), deferred=True)
            )
        for topic in self.topics:
            # pool-subject fan-in (queue group like sdk/runtime/worker.go:106):
            # jobs dispatched to the pool topic land on worker 0's queue
            self._subs.append(
                self.bus.subscribe(topic, self._make_on_job(0), queue_group=f"pool.{self.pool}", deferred=True)
            )
        self._subs.append(self.bus.subscribe(subj.SUBJECT_CANCEL, self._on_cancel))
        self.send_heartbeats()
        return self

    def stop(self) -> None:
        for s in self._subs:
            s.unsubscribe()

    # -- heartbeats ----------------------------------------------------------------
    def heartbeats(self) -> List[Heartbeat]:
        with self._mu:
            active = list(self._active)
        return [
            Heartbeat(
                worker_id=wid,
                type="gpu-pool",
                active_jobs=active[i],
                pool=self.pool,
                max_parallel_jobs=self.max_parallel_jobs,
            )
            for i, wid in enumerate(self.worker_ids)
        ]

    def send_heartbeats(self) -> None:
        for hb in self.heartbeats():
            self.bus.publish(subj.SUBJECT_HEARTBEAT, BusPacket(heartbeat=hb))

    # -- intake ----------------------------------------------------------------------
    def _make_on_job(self, widx: int):
        def on_job(subject: str, pkt: BusPacket) -> None:
            req = pkt.job_request
            if req is None or not req.job_id:
                return
            with self._mu:
                self._staged.append((req, pkt.trace_id, widx))
                self._active[widx] += 1

        return on_job

    def _on_cancel(self, subject: str, pkt: BusPacket) -> None:
        if pkt.job_cancel is not None and pkt.job_cancel.job_id:
            with self._mu:
                self._cancelled.add(pkt.job_cancel.job_id)

    def staged_count(self) -> int:
        with self._mu:
            return len(self._staged)

    # -- batched device execution -------------------------------------------------------
    def execute(self) -> int:
        """Run every staged job through the device echo kernel in one batch.
        Returns jobs completed (results published on sys.job.result)."""
        with self._mu:
            batch = self._staged
            self._staged = []
        if not batch:
            return 0
        start = self.clock.now()

        jobs: List[Tuple[JobRequest, str, int, bytes]] = []
        for req, trace_id, widx in batch:
            if req.job_id in self._cancelled:
                with self._mu:
                    self._cancelled.discard(req.job_id)
                self._publish(req, trace_id, widx, JobStatus.CANCELLED, "", "cancelled",
                              "job cancelled", start)
                continue
            blob = b""
            if req.context_ptr:
                try:
                    blob = self.memory.get_pointer(req.context_ptr) or b""
                except ValueError:
                    blob = b""
            if len(blob) > self.max_payload_bytes:
                self._publish(req, trace_id, widx, JobStatus.FAILED, "", "payload_too_large",
                              f"context {len(blob)} B > {self.max_payload_bytes} B", start)
                continue
            jobs.append((req, trace_id, widx, blob))

        if jobs:
            K = len(jobs)
            stride = max(1, (max(len(b) for (_, _, _, b) in jobs) + 3) // 4)
            arena = torch.zeros(K * stride * 4, dtype=torch.uint8)
            for k, (_, _, _, blob) in enumerate(jobs):
                if blob:
                    arena[k * stride * 4: k * stride * 4 + len(blob)] = torch.frombuffer(
                        bytearray(blob), dtype=torch.uint8
                    )
            ctx = arena.view(torch.int32).to(self.device)
            res = torch.zeros_like(ctx)
            sums = torch.zeros(K, dtype=torch.int32, device=self.device)
            slots = torch.arange(K, dtype=torch.int32, device=self.device)
            self.ext.echo_execute_indexed(ctx, slots, res, sums, stride)
            res_host = res.cpu().view(torch.uint8)  # result bytes round-trip through HBM
            for k, (req, trace_id, widx, blob) in enumerate(jobs):
                payload = bytes(res_host[k * stride * 4: k * stride * 4 + len(blob)]) if blob else b"{}"
                ptr = self.memory.put_result(req.job_id, payload)
                self._publish(req, trace_id, widx, JobStatus.SUCCEEDED, ptr, "", "", start)
            self.jobs_executed += K
            self.batches_executed += 1

        with self._mu:
            for req, _, widx in batch:
                self._active[widx] = max(0, self._active[widx] - 1)
        return len(jobs)

    # -- result publish -------------------------------------------------------------------
    def _publish(self, req: JobRequest, trace_id: str, widx: int, status: JobStatus,
                 result_ptr: str, code: str, msg: str, start: float) -> None:
        res = JobResult(
            job_id=req.job_id,
            status=status,
            result_ptr=result_ptr,
            worker_id=self.worker_ids[widx],
            execution_ms=int((self.clock.now() - start) * 1000),
            error_code=code,
            error_message=msg,
        )
        self.bus.publish(subj.SUBJECT_RESULT, BusPacket(trace_id=trace_id, job_result=res))
