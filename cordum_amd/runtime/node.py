"""Single-process control-plane node: the 5 reference microservices fused.

The reference runs gateway / scheduler / safety-kernel / workflow-engine /
context-engine as separate binaries over NATS+Redis. The
MI355X design collapses them into one process per node (BASELINE.json north
star): this object owns the bus, every store, the safety kernel, the
scheduler engine, the workflow engine, worker runtimes and the reconcilers,
and exposes a tick() the control loop (or the GPU pipeline driver) calls.

The HTTP/gRPC gateway (gateway/app.py) and the GPU data plane
(ops/pipeline.py) are both built on top of this object.
"""
from __future__ import annotations

import json
from typing import List, Optional

from ..bus import LoopbackBus
from ..protocol import subjects as subj
from ..protocol.capv2 import BusPacket, JobRequest
from ..safety import SafetyKernel, parse_safety_policy
from ..scheduler import (
    Engine as SchedulerEngine,
    LeastLoadedStrategy,
    PendingReplayer,
    PoolRouting,
    Reconciler,
    SafetyChecker,
    WorkerRegistry,
    routing_from_pools_yaml,
)
from ..store import (
    ArtifactStore,
    ConfigService,
    DLQEntry,
    DLQStore,
    JobStore,
    LockService,
    MemoryStore,
    SchemaRegistry,
)
from ..utils.clock import Clock, SYSTEM_CLOCK
from ..utils.metrics import Metrics as PromMetrics
from ..workflow import Engine as WorkflowEngine, RunReconciler, WorkflowService, WorkflowStore
from .worker import Worker, echo_handler

DEFAULT_ROUTING = PoolRouting(topics={"job.default": ["default"]}, pools={})


class Node:
    def __init__(
        self,
        clock: Clock = SYSTEM_CLOCK,
        routing: Optional[PoolRouting] = None,
        policy_yaml: str = "",
        safety_cache_ttl_s: float = 30.0,
        dispatch: str = "host",
        device=None,
        backend: Optional[str] = None,
    ):
        """dispatch="device" runs the safety gate and worker scoring through
        the K1/K2 kernels (runtime/device_dispatch.py): on a GPU host with the
        HIP extension (backend="ext", the default there), on CPU via the torch
        reference ops (backend="ref") — same code path, same semantics."""
        self.clock = clock
        self.bus = LoopbackBus(clock=clock)
        self.job_store = JobStore(clock=clock)
        self.memory = MemoryStore(clock=clock)
        self.artifacts = ArtifactStore(self.memory, clock=clock)
        self.dlq = DLQStore(clock=clock)
        self.locks = LockService(clock=clock)
        self.schemas = SchemaRegistry()
        self.configsvc = ConfigService()
        self.registry = WorkerRegistry(clock=clock)

        base_policy = parse_safety_policy(policy_yaml) if policy_yaml else None
        self.safety_kernel = SafetyKernel(
            base_policy, configsvc=self.configsvc, cache_ttl_s=safety_cache_ttl_s, clock=clock
        )
        self.strategy = LeastLoadedStrategy(routing or DEFAULT_ROUTING)
        self.metrics_exporter = PromMetrics()
        self.dispatch_mode = dispatch
        self.device = None
        self._ext = None
        self.device_gate = None
        self.device_worker_table = None
        self.device_pools: List = []
        if dispatch == "device":
            import torch

            from ..ops.worker_table import DeviceWorkerTable
            from .device_dispatch import DeviceDispatchEngine
            from .device_gate import DeviceBatchGate

            if device is None:
                device = "cuda:0" if torch.cuda.is_available() else "cpu"
            self.device = torch.device(device)
            backend = backend or ("ext" if self.device.type == "cuda" else "ref")
            if backend == "ref":
                from ..ops.pipeline import _RefOps

                self._ext = _RefOps()
            else:
                from ..ops import get_ext

                self._ext = get_ext(required=True)
            self.device_gate = DeviceBatchGate(self.safety_kernel, self.device, self._ext)
            self.device_worker_table = DeviceWorkerTable(self.device, self._ext)
            self.scheduler = DeviceDispatchEngine(
                self.bus,
                self.job_store,
                SafetyChecker(self.safety_kernel, clock=clock),
                self.registry,
                self.strategy,
                configsvc=self.configsvc,
                metrics=self.metrics_exporter,
                clock=clock,
                gate=self.device_gate,
                worker_table=self.device_worker_table,
            )
        else:
            self.scheduler = SchedulerEngine(
                self.bus,
                self.job_store,
                SafetyChecker(self.safety_kernel, clock=clock),
                self.registry,
                self.strategy,
                configsvc=self.configsvc,
                metrics=self.metrics_exporter,
                clock=clock,
            )
        self.workflow_store = WorkflowStore(clock=clock)
        self.workflow = WorkflowEngine(
            self.workflow_store,
            self.bus,
            memory=self.memory,
            config=self.configsvc,
            schema_registry=self.schemas,
            clock=clock,
        )
        self.workflow_service = WorkflowService(self.workflow, self.bus)
        self.scheduler_reconciler = Reconciler(self.job_store, clock=clock)
        self.pending_replayer = PendingReplayer(self.scheduler, self.job_store, clock=clock)
        self.run_reconciler = RunReconciler(self.workflow, self.workflow_store, self.job_store, clock=clock)
        self.workers: List[Worker] = []
        self.wal = None  # optional Checkpointer: submissions are WAL-appended
        self._started = False
        self._route_config_watch()

    # -- lifecycle -------------------------------------------------------------
    def start(self) -> "Node":
        if self._started:
            return self
        self.scheduler.start()
        self.workflow_service.start()
        self.bus.subscribe(subj.SUBJECT_DLQ, self._dlq_tap)
        self._started = True
        return self

    def add_worker(
        self,
        worker_id: str,
        handler=echo_handler,
        topics: Optional[List[str]] = None,
        pool: str = "default",
        **kw,
    ) -> Worker:
        w = Worker(
            bus=self.bus,
            memory=self.memory,
            worker_id=worker_id,
            handler=handler,
            topics=topics or ["job.default"],
            pool=pool,
            clock=self.clock,
            **kw,
        )
        w.start()
        self.workers.append(w)
        return w

    def add_device_worker_pool(self, n_workers: int = 4, pool: str = "default",
                               topics: Optional[List[str]] = None, **kw):
        """GPU-resident echo worker pool (ops/worker_pool.py): jobs routed to
        its worker subjects execute as one batched device kernel."""
        assert self.dispatch_mode == "device", "device pool needs dispatch='device'"
        from ..ops.worker_pool import DeviceWorkerPool

        p = DeviceWorkerPool(
            self.bus, self.memory, self.device, self._ext,
            n_workers=n_workers, pool=pool, topics=topics or ["job.default"],
            clock=self.clock, **kw,
        ).start()
        self.device_pools.append(p)
        return p

    # -- ticking -----------------------------------------------------------------
    def _pump_dispatch(self) -> int:
        """One bus pump + device flush + device-pool batch execution round."""
        n = self.bus.pump()
        flush = getattr(self.scheduler, "flush", None)
        if flush is not None:
            n += flush()
        for p in self.device_pools:
            n += p.execute()
        return n

    def tick(self) -> None:
        """One control-loop iteration: drain bus queues, fire timers,
        heartbeats, reconcilers."""
        for w in self.workers:
            w.send_heartbeat()
        for p in self.device_pools:
            p.send_heartbeats()
        self._pump_dispatch()
        self.workflow.pump_timers()
        self._pump_dispatch()

    def reconcile(self) -> None:
        self.scheduler_reconciler.tick()
        self.pending_replayer.tick()
        self.run_reconciler.tick()
        self.write_worker_snapshot()
        self._pump_dispatch()

    def write_worker_snapshot(self) -> None:
        """Periodic cluster snapshot under `sys:workers:snapshot`
        (cmd/cordum-scheduler/main.go:128-162, 5s writer)."""
        snap = self.registry.cluster_snapshot()
        self.memory.put("sys:workers:snapshot", json.dumps(snap).encode())

    def drain(self, max_iters: int = 64) -> None:
        """Pump until quiescent at the current clock (tests / sync callers)."""
        for _ in range(max_iters):
            n = self._pump_dispatch()
            n += self.workflow.pump_timers()
            if n == 0:
                return

    # -- submit (the gateway submit path without HTTP) ------------------------------
    def submit_job(self, req: JobRequest, trace_id: str = "", context: Optional[bytes] = None) -> str:
        from ..utils.ids import new_trace_id

        trace_id = trace_id or new_trace_id()
        if context is not None:
            req.context_ptr = self.memory.put_context(req.job_id, context)
        if self.wal is not None:
            self.wal.wal_append(req, trace_id)
        self.bus.publish(subj.SUBJECT_SUBMIT, BusPacket(trace_id=trace_id, job_request=req))
        return trace_id

    # -- internals --------------------------------------------------------------------
    def _dlq_tap(self, subject: str, pkt: BusPacket) -> None:
        """gateway.go:551-601: DLQ tap persists entries + synthesizes state."""
        res = pkt.job_result
        if res is None or not res.job_id:
            return
        meta = self.job_store.get_job_meta(res.job_id)
        self.dlq.add(
            DLQEntry(
                job_id=res.job_id,
                topic=meta.get("topic", ""),
                status=res.status.name if hasattr(res.status, "name") else str(res.status),
                reason=res.error_message,
                reason_code=res.error_code,
                last_state=meta.get("state", ""),
                attempts=int(meta.get("attempts", 0) or 0),
                tenant=meta.get("tenant", ""),
                trace_id=meta.get("trace_id", ""),
            )
        )

    def _route_config_watch(self) -> None:
        """Scheduler config overlay: hot-swap routing when cfg:system:default
        pools change (cmd/cordum-scheduler/config_overlay.go:113-153)."""

        def on_write(scope: str, doc_id: str) -> None:
            if scope != "system" or doc_id != "default":
                return
            doc = self.configsvc.get("system", "default") or {}
            pools_doc = doc.get("pools")
            if isinstance(pools_doc, dict):
                self.strategy.update_routing(routing_from_pools_yaml(pools_doc))
            timeouts = doc.get("timeouts")
            if isinstance(timeouts, dict):
                self.scheduler_reconciler.update_timeouts(
                    float(timeouts.get("dispatch_timeout_sec", 0) or 0),
                    float(timeouts.get("running_timeout_sec", 0) or 0),
                )

        self.configsvc.watch(on_write)
