"""Workflow engine: the fan-out/retry/approval/delay state machine.

Semantics oracle: core/workflow/engine.go —
 StartRun :67-82, RerunFrom :85-151 (dep-output cloning, dry-run),
 HandleJobResult :154-299 (job-id `runID:stepID@attempt` parse, dedup by
 job-id + processed-state, retry w/ exponential backoff, output schema
 validation, ≤256 KiB result inlining into run context `steps.<id>`,
 for_each child aggregation, run-status recompute, next-wave scheduling),
 ApproveStep :302-346, CancelRun :349-451 (cancel steps + JobCancel per
 in-flight job), scheduleReady :453-827 (deps gate, condition gate/step,
 approval → WAITING, delay timers, notify alerts, for_each fan-out with
 max_parallel window and child ids `step[idx]`, worker dispatch via
 sys.job.submit), templates :873-964, applyResult :1524-1560,
 computeBackoff :1573-1595, aggregateChildren :1623-1645,
 updateRunStatus :1647-1699, timeline :1789-1809.

Differences from the reference (documented):
 - steps iterate in sorted-key order (Go map order is unspecified);
 - delay/backoff timers go into an internal heap drained by `pump_timers()`
   from the node control loop instead of goroutines;
 - run documents are owned by this process (single-node), so the run-level
   Redis lock becomes the engine mutex.
"""
from __future__ import annotations

import heapq
import json
import threading
from typing import Any, Callable, Dict, List, Optional, Tuple

from ..bus import Bus
from ..protocol import subjects as subj
from ..protocol.capv2 import (
    ActorType,
    Budget,
    BusPacket,
    JobCancel,
    JobMetadata,
    JobPriority,
    JobRequest,
    JobResult,
    JobStatus,
    SystemAlert,
)
from ..store.memory_store import MemoryStore, pointer_for_key
from ..store.schema_registry import SchemaRegistry, validate_value
from ..utils.clock import Clock, SYSTEM_CLOCK
from . import eval as wfeval
from .models import (
    RUN_CANCELLED,
    RUN_FAILED,
    RUN_PENDING,
    RUN_RUNNING,
    RUN_SUCCEEDED,
    RUN_TERMINAL,
    RUN_TIMED_OUT,
    RUN_WAITING,
    STEP_CANCELLED,
    STEP_FAILED,
    STEP_PENDING,
    STEP_RUNNING,
    STEP_SUCCEEDED,
    STEP_TERMINAL,
    STEP_TIMED_OUT,
    STEP_WAITING,
    Step,
    StepRun,
    TimelineEvent,
    Workflow,
    WorkflowRun,
)
from .store import WorkflowStore

MAX_INLINE_RESULT_BYTES = 256 << 10


class Engine:
    def __init__(
        self,
        store: WorkflowStore,
        bus: Bus,
        memory: Optional[MemoryStore] = None,
        config=None,  # ConfigService-like with .effective(org, team, workflow, step)
        schema_registry: Optional[SchemaRegistry] = None,
        clock: Clock = SYSTEM_CLOCK,
    ):
        self.store = store
        self.bus = bus
        self.memory = memory
        self.config = config
        self.schema_registry = schema_registry
        self.clock = clock
        self._mu = threading.RLock()
        self._timers: List[Tuple[float, int, str, str]] = []  # (due, seq, wf_id, run_id)
        self._timer_seq = 0
        self.on_step_dispatched: Optional[Callable[[str, str, str], None]] = None
        self.on_step_finished: Optional[Callable[[str, str, str], None]] = None

    # -- public API ---------------------------------------------------------
    def start_run(self, workflow_id: str, run_id: str) -> None:
        with self._mu:
            wf = self.store.get_workflow(workflow_id)
            run = self.store.get_run(run_id)
            if run.status in RUN_TERMINAL:
                return
            self._schedule_ready(wf, run)

    def rerun_from(self, run_id: str, step_id: str = "", dry_run: bool = False) -> str:
        from ..utils.ids import new_id

        with self._mu:
            if not run_id:
                raise ValueError("run id required")
            run = self.store.get_run(run_id)
            wf = self.store.get_workflow(run.workflow_id)
            deps: set = set()
            if step_id:
                if step_id not in wf.steps:
                    raise ValueError("step not found")
                _collect_dependencies(wf, step_id, deps)
            now = self.clock.now()
            new_run = WorkflowRun(
                id=new_id(),
                workflow_id=run.workflow_id,
                org_id=run.org_id,
                team_id=run.team_id,
                input=json.loads(json.dumps(run.input)),
                context=_clone_context_for_deps(run.context, deps),
                status=RUN_PENDING,
                triggered_by=run.triggered_by,
                created_at=now,
                updated_at=now,
                rerun_of=run.id,
                rerun_step=step_id,
                dry_run=dry_run,
                labels=dict(run.labels),
                metadata=dict(run.metadata),
            )
            new_run.metadata["rerun_of"] = run.id
            if step_id:
                new_run.metadata["rerun_step"] = step_id
            if dry_run:
                new_run.metadata["dry_run"] = "true"
                new_run.labels["dry_run"] = "true"
            for dep in deps:
                prev = run.steps.get(dep)
                if prev is None or prev.status != STEP_SUCCEEDED:
                    raise ValueError(f"dependency {dep} not succeeded")
                new_run.steps[dep] = prev.clone()
            self.store.create_run(new_run)
            return new_run.id

    def handle_job_result(self, res: JobResult) -> None:
        if res is None or not res.job_id:
            return
        run_id, step_id = split_job_id(res.job_id)
        if not run_id or not step_id:
            return
        with self._mu:
            try:
                run = self.store.get_run(run_id)
            except KeyError:
                return
            if run.status in RUN_TERMINAL:
                return
            try:
                wf = self.store.get_workflow(run.workflow_id)
            except KeyError:
                return

            prev_status = run.status
            base_step_id, child_key = split_for_each_step(step_id)
            step_def = wf.steps.get(base_step_id)
            now = self.clock.now()
            attempt = parse_attempt(res.job_id)

            if child_key:
                parent = run.steps.get(base_step_id) or StepRun(step_id=base_step_id)
                child = parent.children.get(step_id) or run.steps.get(step_id) or StepRun(step_id=step_id)
                if child.job_id and child.job_id != res.job_id:
                    return
                if not child.job_id:
                    child.job_id = res.job_id
                if attempt > child.attempts:
                    child.attempts = attempt
                if child.job_id == res.job_id and _should_ignore(child):
                    return
                retry, delay = self._apply_result(child, res, step_def, now)
                self._post_result(run, child, step_def, res, retry, step_id, now, apply_output_path=False)
                parent.children[step_id] = child
                run.steps[step_id] = child
                parent.status = aggregate_children(parent)
                if parent.status in (STEP_SUCCEEDED, STEP_FAILED):
                    parent.completed_at = now
                run.steps[base_step_id] = parent
                if retry and delay > 0:
                    self._schedule_after(delay, run.workflow_id, run.id)
                self._maybe_on_finished(run, step_id, child, retry)
            else:
                sr = run.steps.get(step_id)
                if sr is not None and sr.job_id and sr.job_id != res.job_id:
                    return
                if sr is None:
                    sr = StepRun(step_id=step_id)
                if not sr.job_id:
                    sr.job_id = res.job_id
                if attempt > sr.attempts:
                    sr.attempts = attempt
                if sr.job_id == res.job_id and _should_ignore(sr):
                    return
                retry, delay = self._apply_result(sr, res, step_def, now)
                self._post_result(run, sr, step_def, res, retry, step_id, now, apply_output_path=True)
                run.steps[step_id] = sr
                if retry and delay > 0:
                    self._schedule_after(delay, run.workflow_id, run.id)
                self._maybe_on_finished(run, step_id, sr, retry)

            run.updated_at = now
            update_run_status(run, wf, now)
            if prev_status != run.status:
                self._timeline(run, "run_status", status=run.status)
            self.store.update_run(run)
            if run.status == RUN_RUNNING:
                self._schedule_ready(wf, run)

    def approve_step(self, run_id: str, step_id: str, approved: bool) -> None:
        with self._mu:
            run = self.store.get_run(run_id)
            wf = self.store.get_workflow(run.workflow_id)
            sr = run.steps.get(step_id)
            if sr is None:
                raise ValueError("step not found")
            if sr.status != STEP_WAITING:
                raise ValueError("step not waiting")
            now = self.clock.now()
            prev_status = run.status
            sr.status = STEP_SUCCEEDED if approved else STEP_FAILED
            sr.completed_at = now
            run.steps[step_id] = sr
            update_run_status(run, wf, now)
            self._timeline(run, "step_approved" if approved else "step_rejected", step_id=step_id, status=sr.status)
            if prev_status != run.status:
                self._timeline(run, "run_status", status=run.status)
            self.store.update_run(run)
            if approved and run.status == RUN_RUNNING:
                self._schedule_ready(wf, run)

    def cancel_run(self, run_id: str) -> None:
        with self._mu:
            run = self.store.get_run(run_id)
            wf = self.store.get_workflow(run.workflow_id)
            now = self.clock.now()
            cancel_job_ids: List[str] = []
            for step_id in wf.steps:
                sr = run.steps.get(step_id) or StepRun(step_id=step_id)
                cancel_job_ids.extend(_collect_cancelable_jobs(sr))
                _cancel_step_run(sr, now)
                run.steps[step_id] = sr
            for sid, sr in list(run.steps.items()):
                cancel_job_ids.extend(_collect_cancelable_jobs(sr))
                _cancel_step_run(sr, now)
                run.steps[sid] = sr
            run.status = RUN_CANCELLED
            run.completed_at = now
            run.updated_at = now
            self.store.update_run(run)
            self._timeline(run, "run_status", status=run.status, message="run cancelled")
            for job_id in dict.fromkeys(j for j in cancel_job_ids if j):
                self._publish_job_cancel(job_id, "workflow run cancelled")

    # -- timers ---------------------------------------------------------------
    def _schedule_after(self, delay_s: float, workflow_id: str, run_id: str) -> None:
        self._timer_seq += 1
        heapq.heappush(self._timers, (self.clock.now() + delay_s, self._timer_seq, workflow_id, run_id))

    def pump_timers(self) -> int:
        """Fire due delay/backoff timers (engine.go:1779-1787 scheduleAfter)."""
        n = 0
        while True:
            with self._mu:
                if not self._timers or self._timers[0][0] > self.clock.now():
                    return n
                _, _, wf_id, run_id = heapq.heappop(self._timers)
            try:
                self.start_run(wf_id, run_id)
                n += 1
            except KeyError:
                pass

    def next_timer_due(self) -> Optional[float]:
        with self._mu:
            return self._timers[0][0] if self._timers else None

    # -- scheduleReady ----------------------------------------------------------
    def _schedule_ready(self, wf: Workflow, run: WorkflowRun) -> None:
        """One scheduleReady call = sweep to fixpoint: steps that complete
        inline (condition, skip, notify, zero-delay, empty for_each) unlock
        their dependents within the same call. The reference gets the same
        effect across reconciler re-drives; sweeping to fixpoint removes the
        Go map-iteration-order dependence."""
        if run.status in RUN_TERMINAL:
            return
        now = self.clock.now()
        prev_status = run.status
        if run.status == RUN_PENDING:
            run.status = RUN_RUNNING
            run.started_at = now

        for _ in range(len(wf.steps) + 1):
            if not self._schedule_pass(wf, run, now):
                break

        update_run_status(run, wf, now)
        if prev_status != run.status:
            self._timeline(run, "run_status", status=run.status)
        run.updated_at = now
        self.store.update_run(run)

    def _schedule_pass(self, wf: Workflow, run: WorkflowRun, now: float) -> bool:
        progressed = False
        for step_id in sorted(wf.steps):
            step = wf.steps[step_id]
            sr = run.steps.get(step_id) or StepRun(step_id=step_id)
            if sr.status and sr.status not in (STEP_PENDING, STEP_WAITING):
                # for_each steps stay RUNNING while children need dispatch;
                # delay steps re-checked while RUNNING
                if not step.for_each:
                    if step.type != "delay":
                        continue
                elif sr.status != STEP_RUNNING:
                    continue
            if not deps_satisfied(step, run):
                continue

            scope = _build_scope(run, None)

            # condition gate for non-condition steps
            if step.condition and step.type != "condition":
                try:
                    ok = wfeval.eval_condition(step.condition, scope)
                except wfeval.EvalError:
                    continue
                if not ok:
                    sr.status = STEP_SUCCEEDED
                    sr.started_at = now
                    sr.completed_at = now
                    run.steps[step_id] = sr
                    progressed = True
                    continue

            if step.type == "condition":
                if not sr.status or sr.status == STEP_PENDING:
                    sr.status = STEP_RUNNING
                    sr.started_at = now
                    cond = step.condition.strip()
                    if not cond:
                        self._fail_step(run, sr, step_id, "condition expression required", "step_condition_failed", now)
                        continue
                    try:
                        ok = wfeval.eval_condition(cond, scope)
                    except wfeval.EvalError as e:
                        self._fail_step(run, sr, step_id, str(e), "step_condition_failed", now)
                        continue
                    err = self._validate_inline_output(step, ok)
                    if err:
                        self._fail_step(run, sr, step_id, err, "step_condition_failed", now)
                        continue
                    sr.status = STEP_SUCCEEDED
                    sr.completed_at = now
                    sr.output = ok
                    run.steps[step_id] = sr
                    progressed = True
                    _record_inline_output(run, step_id, step, ok)
                    self._timeline(run, "step_condition_evaluated", step_id=step_id, status=sr.status, data={"value": ok})
                    self._maybe_on_finished(run, step_id, sr, retry=False)
                run.steps[step_id] = sr
                continue

            if step.type == "approval":
                if not sr.status or sr.status == STEP_PENDING:
                    sr.status = STEP_WAITING
                    sr.started_at = now
                    run.status = RUN_WAITING
                    self._timeline(run, "step_waiting", step_id=step_id, status=sr.status, message="approval requested")
                run.steps[step_id] = sr
                continue

            if step.type == "delay":
                try:
                    delay = _delay_for_step(step, now)
                except ValueError as e:
                    self._fail_step(run, sr, step_id, str(e), "step_delay_failed", now)
                    continue
                if not sr.status or sr.status == STEP_PENDING:
                    sr.status = STEP_RUNNING
                    sr.started_at = now
                    if delay <= 0:
                        sr.status = STEP_SUCCEEDED
                        sr.completed_at = now
                        sr.next_attempt_at = None
                        run.steps[step_id] = sr
                        progressed = True
                        self._timeline(run, "step_delay_completed", step_id=step_id, status=sr.status, message="delay completed")
                        self._maybe_on_finished(run, step_id, sr, retry=False)
                        continue
                    sr.next_attempt_at = now + delay
                    run.steps[step_id] = sr
                    self._timeline(run, "step_delay_started", step_id=step_id, status=sr.status,
                                   message="delay started", data={"delay_ms": int(delay * 1000)})
                    self._schedule_after(delay, run.workflow_id, run.id)
                    continue
                if sr.status == STEP_RUNNING and sr.next_attempt_at is not None and sr.next_attempt_at <= now:
                    sr.status = STEP_SUCCEEDED
                    sr.completed_at = now
                    sr.next_attempt_at = None
                    run.steps[step_id] = sr
                    progressed = True
                    self._timeline(run, "step_delay_completed", step_id=step_id, status=sr.status, message="delay completed")
                    self._maybe_on_finished(run, step_id, sr, retry=False)
                    continue
                run.steps[step_id] = sr
                continue

            if step.type == "notify":
                if not sr.status or sr.status == STEP_PENDING:
                    try:
                        payload = wfeval.eval_templates(step.input, scope)
                    except wfeval.EvalError as e:
                        self._fail_step(run, sr, step_id, str(e), "step_event_failed", now)
                        continue
                    alert = _build_event_alert(step, payload, run)
                    pkt = BusPacket(trace_id=run.id, protocol_version=1, alert=alert)
                    self.bus.publish(subj.SUBJECT_WORKFLOW_EVENT, pkt)
                    sr.status = STEP_SUCCEEDED
                    sr.started_at = now
                    sr.completed_at = now
                    run.steps[step_id] = sr
                    progressed = True
                    self._timeline(run, "step_event_emitted", step_id=step_id, status=sr.status,
                                   message=alert.message, data={"payload": payload})
                    self._maybe_on_finished(run, step_id, sr, retry=False)
                continue

            if step.for_each:
                try:
                    items = wfeval.eval_for_each(step.for_each, scope)
                except wfeval.EvalError:
                    continue
                if not items:
                    if sr.status != STEP_SUCCEEDED:
                        progressed = True
                    sr.status = STEP_SUCCEEDED
                    sr.started_at = now
                    sr.completed_at = now
                    run.steps[step_id] = sr
                    continue
                # pre-create children (engine.go:687-698)
                for idx in range(len(items)):
                    child_id = f"{step_id}[{idx}]"
                    if child_id not in sr.children:
                        child = StepRun(step_id=child_id, status=STEP_PENDING)
                        sr.children[child_id] = child
                        run.steps[child_id] = child
                sr.status = STEP_RUNNING
                if sr.started_at is None:
                    sr.started_at = now
                running_children = sum(1 for c in sr.children.values() if c.status == STEP_RUNNING)
                for idx, item in enumerate(items):
                    if step.max_parallel > 0 and running_children >= step.max_parallel:
                        break
                    child_id = f"{step_id}[{idx}]"
                    child = sr.children.get(child_id) or StepRun(step_id=child_id, status=STEP_PENDING)
                    if child.status and child.status != STEP_PENDING:
                        continue
                    if child.next_attempt_at is not None and child.next_attempt_at > now:
                        continue
                    job_id = f"{run.id}:{child_id}@{child.attempts + 1}"
                    req = self._build_job_request(wf, run, step, child_id, job_id)
                    req.env["foreach_index"] = str(idx)
                    try:
                        req.env["foreach_item"] = json.dumps(item)
                    except (TypeError, ValueError):
                        pass
                    err, payload = self._build_payload(run, step, item)
                    if err:
                        child.status = STEP_FAILED
                        child.error = {"message": err}
                        sr.children[child_id] = child
                        run.steps[child_id] = child
                        continue
                    ptr = self._put_job_context(job_id, payload)
                    if ptr:
                        req.context_ptr = ptr
                    self.bus.publish(subj.SUBJECT_SUBMIT, BusPacket(trace_id=run.id, protocol_version=1, job_request=req))
                    child.status = STEP_RUNNING
                    child.started_at = now
                    child.attempts += 1
                    child.job_id = job_id
                    child.input = payload
                    child.item = item
                    running_children += 1
                    data: Dict[str, Any] = {"foreach_index": idx}
                    if req.context_ptr:
                        data["context_ptr"] = req.context_ptr
                    self._timeline(run, "step_dispatched", step_id=child_id, job_id=job_id, status=child.status, data=data)
                    if self.on_step_dispatched:
                        self.on_step_dispatched(run.id, child_id, job_id)
                    sr.children[child_id] = child
                    run.steps[child_id] = child
                run.steps[step_id] = sr
                continue

            # backoff window
            if sr.next_attempt_at is not None and sr.next_attempt_at > now:
                run.steps[step_id] = sr
                continue

            # worker-class step dispatch
            job_id = f"{run.id}:{step_id}@{sr.attempts + 1}"
            req = self._build_job_request(wf, run, step, step_id, job_id)
            err, payload = self._build_payload(run, step, None)
            if err:
                sr.status = STEP_FAILED
                sr.error = {"message": err}
                run.steps[step_id] = sr
                continue
            ptr = self._put_job_context(job_id, payload)
            if ptr:
                req.context_ptr = ptr
            self.bus.publish(subj.SUBJECT_SUBMIT, BusPacket(trace_id=run.id, protocol_version=1, job_request=req))
            sr.status = STEP_RUNNING
            sr.started_at = now
            sr.attempts += 1
            sr.job_id = job_id
            sr.input = payload
            data = {"context_ptr": req.context_ptr} if req.context_ptr else None
            self._timeline(run, "step_dispatched", step_id=step_id, job_id=job_id, status=sr.status, data=data)
            if self.on_step_dispatched:
                self.on_step_dispatched(run.id, step_id, job_id)
            run.steps[step_id] = sr

        return progressed

    # -- result application -------------------------------------------------------
    def _apply_result(self, sr: StepRun, res: JobResult, step: Optional[Step], now: float) -> Tuple[bool, float]:
        status = res.status
        if status == JobStatus.SUCCEEDED:
            sr.status = STEP_SUCCEEDED
            sr.completed_at = now
            sr.next_attempt_at = None
            if res.result_ptr:
                sr.output = res.result_ptr
            sr.error = None
            return False, 0.0
        if status in (JobStatus.FAILED, JobStatus.DENIED, JobStatus.TIMEOUT):
            if _should_retry(step, sr):
                delay = compute_backoff(step, sr)
                sr.next_attempt_at = now + delay
                sr.status = STEP_PENDING
                sr.error = {"message": res.error_message}
                return True, delay
            sr.status = STEP_TIMED_OUT if status == JobStatus.TIMEOUT else STEP_FAILED
            sr.completed_at = now
            sr.error = {"message": res.error_message}
            return False, 0.0
        if status == JobStatus.CANCELLED:
            sr.status = STEP_CANCELLED
            sr.completed_at = now
            return False, 0.0
        sr.status = STEP_FAILED
        sr.completed_at = now
        sr.error = {"message": f"unexpected status: {status}"}
        return False, 0.0

    def _post_result(
        self,
        run: WorkflowRun,
        sr: StepRun,
        step_def: Optional[Step],
        res: JobResult,
        retry: bool,
        step_id: str,
        now: float,
        apply_output_path: bool,
    ) -> None:
        if not retry and sr.status == STEP_SUCCEEDED and res.result_ptr:
            err = self._validate_step_output(step_def, res.result_ptr)
            if err:
                sr.status = STEP_FAILED
                sr.completed_at = now
                sr.error = {"message": err}
                self._timeline(run, "step_output_invalid", step_id=step_id, job_id=res.job_id,
                               status=sr.status, result_ptr=res.result_ptr, message=err)
        if not retry and sr.status in STEP_TERMINAL:
            self._timeline(run, "step_completed", step_id=step_id, job_id=res.job_id,
                           status=sr.status, result_ptr=res.result_ptr, message=res.error_message)
        if not retry and sr.status == STEP_SUCCEEDED and res.result_ptr:
            self._record_step_output(run, step_id, step_def, res.result_ptr, apply_output_path)

    def _maybe_on_finished(self, run: WorkflowRun, step_id: str, sr: StepRun, retry: bool) -> None:
        if self.on_step_finished and not retry and sr.status in STEP_TERMINAL:
            self.on_step_finished(run.id, step_id, sr.status)

    # -- payload/request builders ----------------------------------------------------
    def _build_payload(self, run: WorkflowRun, step: Step, item: Any) -> Tuple[str, Dict[str, Any]]:
        base: Dict[str, Any] = {}
        if step.input:
            try:
                evaluated = wfeval.eval_templates(step.input, _build_scope(run, item))
            except wfeval.EvalError as e:
                return str(e), {}
            if isinstance(evaluated, dict):
                base = evaluated
            else:
                return f"step input must be object, got {type(evaluated).__name__}", {}
        elif run.input:
            base = run.input
        out = dict(base)
        if item is not None and "item" not in out:
            out["item"] = item
        err = self._validate_step_input(step, out)
        if err:
            return err, {}
        return "", out

    def _put_job_context(self, job_id: str, payload: Dict[str, Any]) -> str:
        if self.memory is None:
            return ""
        data = json.dumps(payload).encode("utf-8")
        key = f"ctx:{job_id}"
        self.memory.put(key, data)
        return pointer_for_key(key)

    def _build_job_request(self, wf: Workflow, run: WorkflowRun, step: Step, step_id: str, job_id: str) -> JobRequest:
        """engine.go:1320-1415."""
        topic = step.topic or f"job.workflow.{wf.id}"
        priority = JobPriority.BATCH
        raw = (run.input or {}).get("priority")
        if isinstance(raw, str):
            p = raw.strip().lower()
            priority = {"critical": JobPriority.CRITICAL, "interactive": JobPriority.INTERACTIVE}.get(p, JobPriority.BATCH)
        memory_id = f"run:{run.id}"
        raw_mem = (run.input or {}).get("memory_id")
        if isinstance(raw_mem, str) and raw_mem.strip():
            memory_id = raw_mem.strip()
        req = JobRequest(
            job_id=job_id,
            topic=topic,
            priority=priority,
            adapter_id=step.worker_id,
            workflow_id=wf.id,
            memory_id=memory_id,
            tenant_id=run.org_id,
            env={
                "workflow_id": wf.id,
                "run_id": run.id,
                "step_id": step_id,
                "tenant_id": run.org_id,
                "team_id": run.team_id,
                "memory_id": memory_id,
                "context_mode": "raw",
            },
            labels={"workflow_id": wf.id, "run_id": run.id, "step_id": step_id},
        )
        if step.worker_id:
            req.labels["worker_id"] = step.worker_id
        for k, v in (step.route_labels or {}).items():
            req.labels[k] = v
        if step.timeout_sec > 0:
            req.budget = Budget(deadline_ms=step.timeout_sec * 1000)
        meta = _build_step_metadata(run, step)
        if meta is not None:
            req.meta = meta
            if not req.principal_id and meta.actor_id:
                req.principal_id = meta.actor_id
        if run.dry_run or run.metadata.get("dry_run") == "true":
            req.env["dry_run"] = "true"
            req.labels["dry_run"] = "true"
        if self.config is not None:
            try:
                snap = self.config.effective(org=run.org_id, team=run.team_id, workflow=wf.id, step=step_id)
                if snap.config:
                    req.env["CORDUM_EFFECTIVE_CONFIG"] = json.dumps(snap.config, sort_keys=True, separators=(",", ":"))
            except Exception:
                pass
        return req

    # -- output recording / validation ---------------------------------------------
    def _record_step_output(self, run: WorkflowRun, step_id: str, step_def: Optional[Step],
                            result_ptr: str, apply_output_path: bool) -> None:
        """recordStepOutput (engine.go:966-997): inline ≤256 KiB results into
        run context `steps.<id>.output` (+ output_path)."""
        steps = run.context.setdefault("steps", {})
        if not isinstance(steps, dict):
            steps = {}
            run.context["steps"] = steps
        entry: Dict[str, Any] = {"result_ptr": result_ptr}
        inline = self._inline_result(result_ptr)
        if inline is not _MISSING:
            entry["output"] = inline
            if apply_output_path and step_def is not None and step_def.output_path.strip():
                _set_context_path(run.context, step_def.output_path.strip(), inline)
        elif apply_output_path and step_def is not None and step_def.output_path.strip():
            _set_context_path(run.context, step_def.output_path.strip(), result_ptr)
        steps[step_id] = entry

    def _inline_result(self, result_ptr: str):
        if self.memory is None:
            return _MISSING
        try:
            blob = self.memory.get_pointer(result_ptr)
        except ValueError:
            return _MISSING
        if blob is None or len(blob) > MAX_INLINE_RESULT_BYTES:
            return _MISSING
        try:
            return json.loads(blob.decode("utf-8"))
        except (ValueError, UnicodeDecodeError):
            return _MISSING

    def _validate_step_input(self, step: Optional[Step], value: Any) -> str:
        if step is None:
            return ""
        if step.input_schema:
            errs = validate_value(step.input_schema, value)
            if errs:
                return "; ".join(errs)
        elif step.input_schema_id.strip():
            if self.schema_registry is None:
                return "schema registry unavailable"
            ok, errs = self.schema_registry.validate_against(step.input_schema_id.strip(), value)
            if not ok:
                return "; ".join(errs)
        return ""

    def _validate_step_output(self, step: Optional[Step], result_ptr: str) -> str:
        if step is None:
            return ""
        if not step.output_schema and not step.output_schema_id.strip():
            return ""
        value = self._inline_result(result_ptr)
        if value is _MISSING:
            return ""  # cannot inline -> skip validation (reference behavior)
        return self._validate_inline_output(step, value)

    def _validate_inline_output(self, step: Optional[Step], value: Any) -> str:
        if step is None:
            return ""
        if step.output_schema:
            errs = validate_value(step.output_schema, value)
            return "; ".join(errs)
        if step.output_schema_id.strip():
            if self.schema_registry is None:
                return "schema registry unavailable"
            ok, errs = self.schema_registry.validate_against(step.output_schema_id.strip(), value)
            if not ok:
                return "; ".join(errs)
        return ""

    # -- misc -------------------------------------------------------------------
    def _fail_step(self, run: WorkflowRun, sr: StepRun, step_id: str, msg: str, event: str, now: float) -> None:
        sr.status = STEP_FAILED
        sr.error = {"message": msg}
        sr.completed_at = now
        run.steps[step_id] = sr
        self._timeline(run, event, step_id=step_id, status=sr.status, message=msg)
        self._maybe_on_finished(run, step_id, sr, retry=False)

    def _publish_job_cancel(self, job_id: str, reason: str) -> None:
        pkt = BusPacket(trace_id=job_id, protocol_version=1, job_cancel=JobCancel(job_id=job_id, reason=reason))
        self.bus.publish(subj.SUBJECT_CANCEL, pkt)

    def _timeline(self, run: WorkflowRun, type_: str, step_id: str = "", job_id: str = "",
                  status: str = "", result_ptr: str = "", message: str = "",
                  data: Optional[Dict[str, Any]] = None) -> None:
        self.store.append_timeline(
            run.id,
            TimelineEvent(
                time=self.clock.now(), type=type_, run_id=run.id, workflow_id=run.workflow_id,
                step_id=step_id, job_id=job_id, status=status, result_ptr=result_ptr,
                message=message, data=data,
            ),
        )


_MISSING = object()


# -- pure helpers (module-level: shared with the device readiness compiler) ----


def deps_satisfied(step: Step, run: WorkflowRun) -> bool:
    """depsSatisfied (engine.go:1231-1242): all deps SUCCEEDED."""
    for dep in step.depends_on:
        sr = run.steps.get(dep)
        if sr is None or sr.status != STEP_SUCCEEDED:
            return False
    return True


def split_job_id(job_id: str):
    """`runID:stepID@attempt` (engine.go:1244-1256)."""
    parts = job_id.split(":")
    if len(parts) < 2:
        return "", ""
    run_id = ":".join(parts[:-1])
    step_id = parts[-1]
    at = step_id.rfind("@")
    if at > 0:
        step_id = step_id[:at]
    return run_id, step_id


def split_for_each_step(step_id: str):
    idx = step_id.find("[")
    if idx == -1:
        return step_id, ""
    return step_id[:idx], step_id


def parse_attempt(job_id: str) -> int:
    at = job_id.rfind("@")
    if at == -1 or at == len(job_id) - 1:
        return 0
    try:
        n = int(job_id[at + 1:].strip())
    except ValueError:
        return 0
    return n if n > 0 else 0


def _should_ignore(sr: StepRun) -> bool:
    if sr.status in STEP_TERMINAL:
        return True
    if sr.status == STEP_PENDING:
        return sr.next_attempt_at is not None
    return False


def _should_retry(step: Optional[Step], sr: StepRun) -> bool:
    if step is None or step.retry is None:
        return False
    if step.retry.max_retries <= 0:
        return False
    return sr.attempts <= step.retry.max_retries


def compute_backoff(step: Optional[Step], sr: StepRun) -> float:
    """computeBackoff (engine.go:1573-1595): initial * mult^(attempt-1), capped."""
    if step is None or step.retry is None:
        return 1.0
    cfg = step.retry
    initial = cfg.initial_backoff_sec if cfg.initial_backoff_sec > 0 else 1
    mult = cfg.multiplier if cfg.multiplier > 1 else 2.0
    attempt = max(sr.attempts, 1)
    delay = float(initial) * (mult ** (attempt - 1))
    if cfg.max_backoff_sec > 0 and delay > cfg.max_backoff_sec:
        delay = float(cfg.max_backoff_sec)
    return delay


def aggregate_children(parent: StepRun) -> str:
    """aggregateChildren (engine.go:1623-1645)."""
    if not parent.children:
        return parent.status
    all_done = True
    has_failed = False
    for child in parent.children.values():
        if child.status in (STEP_FAILED, STEP_CANCELLED, STEP_TIMED_OUT):
            has_failed = True
        elif child.status == STEP_SUCCEEDED:
            pass
        else:
            all_done = False
    if has_failed:
        return STEP_FAILED
    if all_done:
        return STEP_SUCCEEDED
    return STEP_RUNNING


def update_run_status(run: WorkflowRun, wf: Workflow, now: float) -> None:
    """updateRunStatus (engine.go:1647-1699)."""
    if run.status in (RUN_CANCELLED, RUN_TIMED_OUT):
        return
    has_failed = False
    has_timed_out = False
    waiting = False
    all_done = True
    completed = 0
    for step_id in wf.steps:
        sr = run.steps.get(step_id)
        if sr is None:
            all_done = False
            continue
        if sr.status in (STEP_FAILED, STEP_CANCELLED):
            has_failed = True
        elif sr.status == STEP_TIMED_OUT:
            has_timed_out = True
        elif sr.status == STEP_SUCCEEDED:
            completed += 1
        elif sr.status == STEP_WAITING:
            waiting = True
            all_done = False
        else:
            all_done = False
    if has_failed:
        run.status = RUN_FAILED
        run.completed_at = now
        return
    if has_timed_out:
        run.status = RUN_TIMED_OUT
        run.completed_at = now
        return
    if waiting:
        run.status = RUN_WAITING
        return
    if all_done and completed == len(wf.steps):
        run.status = RUN_SUCCEEDED
        run.completed_at = now
        return
    run.status = RUN_RUNNING


def _build_scope(run: WorkflowRun, item: Any) -> Dict[str, Any]:
    steps = run.context.get("steps") if run.context else None
    scope: Dict[str, Any] = {
        "input": run.input or {},
        "ctx": run.context or {},
        "steps": steps if isinstance(steps, dict) else {},
    }
    if item is not None:
        scope["item"] = item
    return scope


def _record_inline_output(run: WorkflowRun, step_id: str, step: Step, output: Any) -> None:
    steps = run.context.setdefault("steps", {})
    if not isinstance(steps, dict):
        steps = {}
        run.context["steps"] = steps
    steps[step_id] = {"output": output}
    if step.output_path.strip():
        _set_context_path(run.context, step.output_path.strip(), output)


def _set_context_path(ctx: Dict[str, Any], path: str, value: Any) -> None:
    parts = path.split(".")
    cur = ctx
    for i, part in enumerate(parts):
        if i == len(parts) - 1:
            cur[part] = value
            return
        nxt = cur.get(part)
        if not isinstance(nxt, dict):
            nxt = {}
            cur[part] = nxt
        cur = nxt


def _delay_for_step(step: Step, now: float) -> float:
    """delay_sec or RFC3339 delay_until (engine.go delayForStep)."""
    if step.delay_until.strip():
        import datetime

        try:
            dt = datetime.datetime.fromisoformat(step.delay_until.strip().replace("Z", "+00:00"))
        except ValueError:
            raise ValueError(f"invalid delay_until {step.delay_until!r}")
        return dt.timestamp() - now
    if step.delay_sec < 0:
        raise ValueError("negative delay_sec")
    return float(step.delay_sec)


def _build_event_alert(step: Step, payload: Any, run: WorkflowRun) -> SystemAlert:
    severity = "info"
    message = ""
    if isinstance(payload, dict):
        severity = str(payload.get("severity", "info") or "info")
        message = str(payload.get("message", "") or "")
    if not message:
        message = f"workflow event from step {step.id}"
    return SystemAlert(
        severity=severity,
        message=message,
        source="workflow-engine",
        labels={"run_id": run.id, "workflow_id": run.workflow_id, "step_id": step.id},
    )


def _build_step_metadata(run: WorkflowRun, step: Step) -> Optional[JobMetadata]:
    sm = step.meta
    if sm is None:
        return None
    meta = JobMetadata(
        actor_id=sm.actor_id.strip(),
        actor_type={"human": ActorType.HUMAN, "service": ActorType.SERVICE}.get(
            sm.actor_type.strip().lower(), ActorType.UNSPECIFIED
        ),
        idempotency_key=sm.idempotency_key.strip(),
        pack_id=sm.pack_id.strip(),
        capability=sm.capability.strip(),
        risk_tags=[t.strip() for t in sm.risk_tags if t.strip()],
        requires=[t.strip() for t in sm.requires if t.strip()],
        labels=dict(sm.labels),
    )
    if (
        not meta.actor_id and meta.actor_type == ActorType.UNSPECIFIED
        and not meta.idempotency_key and not meta.capability
        and not meta.risk_tags and not meta.requires
        and not meta.pack_id and not meta.labels
    ):
        return None
    return meta


def _collect_dependencies(wf: Workflow, step_id: str, deps: set) -> None:
    step = wf.steps.get(step_id)
    if step is None:
        return
    for dep in step.depends_on:
        if dep not in deps:
            deps.add(dep)
            _collect_dependencies(wf, dep, deps)


def _clone_context_for_deps(ctx: Dict[str, Any], deps: set) -> Dict[str, Any]:
    """Keep only dep step outputs in the cloned run context (RerunFrom)."""
    out: Dict[str, Any] = {}
    steps = ctx.get("steps") if ctx else None
    if isinstance(steps, dict) and deps:
        kept = {sid: v for sid, v in steps.items() if sid in deps}
        if kept:
            out["steps"] = json.loads(json.dumps(kept))
    return out


def _collect_cancelable_jobs(sr: StepRun) -> List[str]:
    out = []
    if sr.status in (STEP_RUNNING, STEP_WAITING, STEP_PENDING) and sr.job_id:
        out.append(sr.job_id)
    for child in sr.children.values():
        if child.status in (STEP_RUNNING, STEP_WAITING, STEP_PENDING) and child.job_id:
            out.append(child.job_id)
    return out


def _cancel_step_run(sr: StepRun, now: float) -> None:
    if sr.status not in STEP_TERMINAL:
        sr.status = STEP_CANCELLED
        sr.completed_at = now
    for child in sr.children.values():
        if child.status not in STEP_TERMINAL:
            child.status = STEP_CANCELLED
            child.completed_at = now
