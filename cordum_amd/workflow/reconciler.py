"""Workflow-engine service glue: result subscription + run reconciler.

Oracle: core/controlplane/workflowengine/engine.go:34-143 (subscribe
sys.job.result in queue group `cordum-workflow-engine`, run-level lock then
HandleJobResult) and reconciler.go:20-189 (5s scan over pending/running/
waiting runs, cross-check step jobs against the JobStore and synthesize
JobResults for finished/TIMEOUT jobs, re-StartRun), jobStatusFromState
:160-175.
"""
from __future__ import annotations

from typing import Optional

from ..bus import Bus
from ..protocol import JobState, is_terminal
from ..protocol import subjects as subj
from ..protocol.capv2 import BusPacket, JobResult, JobStatus
from ..store import JobStore
from ..utils.clock import Clock, SYSTEM_CLOCK
from .engine import Engine
from .models import RUN_PENDING, RUN_RUNNING, RUN_WAITING, STEP_RUNNING
from .store import WorkflowStore

QUEUE_GROUP = "cordum-workflow-engine"

_STATE_TO_STATUS = {
    JobState.SUCCEEDED: JobStatus.SUCCEEDED,
    JobState.FAILED: JobStatus.FAILED,
    JobState.CANCELLED: JobStatus.CANCELLED,
    JobState.TIMEOUT: JobStatus.TIMEOUT,
    JobState.DENIED: JobStatus.DENIED,
}


class WorkflowService:
    """Wires the workflow engine to the bus (result consumption)."""

    def __init__(self, engine: Engine, bus: Bus):
        self.engine = engine
        self.bus = bus
        self._subs = []

    def start(self) -> None:
        self._subs.append(
            self.bus.subscribe(subj.SUBJECT_RESULT, self._on_result, queue_group=QUEUE_GROUP)
        )

    def stop(self) -> None:
        for s in self._subs:
            s.unsubscribe()

    def _on_result(self, subject: str, pkt: BusPacket) -> None:
        if pkt.job_result is not None:
            self.engine.handle_job_result(pkt.job_result)


class RunReconciler:
    def __init__(
        self,
        engine: Engine,
        store: WorkflowStore,
        job_store: Optional[JobStore] = None,
        scan_interval_s: float = 5.0,
        clock: Clock = SYSTEM_CLOCK,
    ):
        self.engine = engine
        self.store = store
        self.job_store = job_store
        self.scan_interval_s = scan_interval_s
        self.clock = clock

    def tick(self) -> int:
        """One reconciliation pass; returns number of runs re-driven."""
        n = 0
        for run in self.store.active_runs():
            if run.status not in (RUN_PENDING, RUN_RUNNING, RUN_WAITING):
                continue
            synthesized = False
            if self.job_store is not None:
                for sid, sr in list(run.steps.items()):
                    if sr.status != STEP_RUNNING or not sr.job_id:
                        continue
                    state = self.job_store.get_state(sr.job_id)
                    if is_terminal(state):
                        status = _STATE_TO_STATUS.get(state, JobStatus.FAILED)
                        meta = self.job_store.get_job_meta(sr.job_id)
                        self.engine.handle_job_result(
                            JobResult(
                                job_id=sr.job_id,
                                status=status,
                                result_ptr=meta.get("result_ptr", ""),
                                error_code=meta.get("error_code", ""),
                                error_message=meta.get("error_message", ""),
                            )
                        )
                        synthesized = True
            if not synthesized and run.status in (RUN_PENDING, RUN_RUNNING):
                try:
                    self.engine.start_run(run.workflow_id, run.id)
                    n += 1
                except KeyError:
                    pass
            elif synthesized:
                n += 1
        return n
