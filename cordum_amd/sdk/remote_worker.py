"""External CAP worker runtime — runs in its OWN process and attaches to a
node over the TCP bus bridge, the analog of the reference's NATS worker
(`sdk/runtime/worker.go:20-320`: queue-subscribe pool subjects, per-job
cancel tracking, MaxParallelJobs semaphore, auto JobResult fill, 10 s
heartbeats).

Data transport matches the reference's pointer scheme: context is fetched
from the compat HTTP surface (`GET /api/v1/memory?ptr=`), results are
stored as artifacts (`POST /api/v1/artifacts`) and referenced by pointer in
the published JobResult — the worker never needs Redis or shared memory.
"""
from __future__ import annotations

import threading
import time
from typing import Callable, Dict, List, Optional

from ..bus.tcp_bridge import BridgeClient
from ..protocol import subjects as subj
from ..protocol.capv2 import BusPacket, Heartbeat, JobRequest, JobResult, JobStatus
from .client import Client

Handler = Callable[[JobRequest, Optional[bytes]], bytes]


class RemoteWorker:
    def __init__(
        self,
        worker_id: str,
        handler: Handler,
        topics: List[str],
        bridge_host: str = "127.0.0.1",
        bridge_port: int = 4230,
        api_base: str = "http://127.0.0.1:8080",
        api_key: str = "",
        pool: str = "default",
        max_parallel: int = 4,
        heartbeat_interval_s: float = 10.0,
        capabilities: Optional[List[str]] = None,
    ):
        self.worker_id = worker_id
        self.handler = handler
        self.topics = topics
        self.pool = pool
        self.max_parallel = max_parallel
        self.heartbeat_interval_s = heartbeat_interval_s
        self.capabilities = capabilities or []
        self.bus = BridgeClient(bridge_host, bridge_port)
        self.api = Client(base_url=api_base, api_key=api_key,
                          principal_id=worker_id)
        from concurrent.futures import ThreadPoolExecutor

        self._sem = threading.Semaphore(max_parallel)
        self._pool = ThreadPoolExecutor(max_workers=max_parallel)
        self._active: Dict[str, bool] = {}
        self._cancelled: Dict[str, bool] = {}
        self._mu = threading.Lock()
        self._stop = threading.Event()

    # -- wiring ---------------------------------------------------------------
    def start(self) -> "RemoteWorker":
        for t in self.topics:
            # SAME queue-group convention as the in-process worker runtime
            # (pool.<pool>, worker.go:96-111): external and in-process
            # members of a pool load-balance together — a different group
            # name would DOUBLE-deliver pool-topic fallback publishes
            self.bus.subscribe(t, queue_group=f"pool.{self.pool}")
        self.bus.subscribe(subj.worker_subject(self.worker_id))
        self.bus.subscribe(subj.SUBJECT_CANCEL)
        self.send_heartbeat()
        threading.Thread(target=self._heartbeat_loop, daemon=True).start()
        threading.Thread(target=self._recv_loop, daemon=True).start()
        return self

    def run_forever(self) -> None:
        self.start()
        while not self._stop.is_set():
            time.sleep(0.2)

    def stop(self) -> None:
        self._stop.set()
        self.bus.close()

    # -- heartbeats (worker.go:239-252) ---------------------------------------
    def send_heartbeat(self) -> None:
        with self._mu:
            active = len(self._active)
        hb = Heartbeat(worker_id=self.worker_id, type="remote", pool=self.pool,
                       active_jobs=active, max_parallel_jobs=self.max_parallel,
                       capabilities=list(self.capabilities))
        self.bus.publish(subj.SUBJECT_HEARTBEAT,
                         BusPacket(protocol_version=1, heartbeat=hb))

    def _heartbeat_loop(self) -> None:
        while not self._stop.is_set():
            time.sleep(self.heartbeat_interval_s)
            try:
                self.send_heartbeat()
            except OSError:
                return

    # -- job loop --------------------------------------------------------------
    def _recv_loop(self) -> None:
        while not self._stop.is_set():
            msg = self.bus.next_message()
            if msg is None:
                return
            subject, pkt = msg
            if pkt.job_cancel is not None:
                with self._mu:
                    if pkt.job_cancel.job_id in self._active:
                        self._cancelled[pkt.job_cancel.job_id] = True
                continue
            if pkt.job_request is None:
                continue
            req = pkt.job_request
            trace = pkt.trace_id
            self._pool.submit(self._run_job, req, trace)

    def _run_job(self, req: JobRequest, trace_id: str) -> None:
        with self._sem:
            with self._mu:
                self._active[req.job_id] = True
            started = time.perf_counter()
            status = JobStatus.SUCCEEDED
            result_ptr = ""
            error_code = error_message = ""
            try:
                ctx = None
                if req.context_ptr:
                    ctx = self.api.memory(req.context_ptr)
                    if not isinstance(ctx, (bytes, bytearray)):
                        import json as _json

                        ctx = _json.dumps(ctx).encode()
                out = self.handler(req, ctx)
                with self._mu:
                    if self._cancelled.pop(req.job_id, False):
                        status = JobStatus.CANCELLED
                if status == JobStatus.SUCCEEDED and out is not None:
                    result_ptr = self.api.artifacts_put(out)["ptr"]
            except Exception as e:  # auto result fill (worker.go:150-204)
                status = JobStatus.FAILED
                error_code = "handler_error"
                error_message = str(e)
            finally:
                with self._mu:
                    self._active.pop(req.job_id, None)
                    self._cancelled.pop(req.job_id, None)
            res = JobResult(
                job_id=req.job_id, status=status, result_ptr=result_ptr,
                worker_id=self.worker_id,
                execution_ms=int((time.perf_counter() - started) * 1000),
                error_code=error_code, error_message=error_message,
            )
            try:
                self.bus.publish(subj.SUBJECT_RESULT,
                                 BusPacket(trace_id=trace_id, protocol_version=1,
                                           job_result=res))
                self.send_heartbeat()
            except OSError:
                pass
