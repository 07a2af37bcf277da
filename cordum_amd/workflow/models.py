"""Workflow data model.

Oracle: core/workflow/models.go:8-171 — 16 step types (worker/approval/
condition/delay/notify + for_each have engine logic), Workflow/Step/
WorkflowRun/StepRun/TimelineEvent shapes, retry config.

Runs are plain dataclasses on the host; the batched readiness sweep
(ops/wf_pipeline.py, K3-WF) packs active runs' dependency bitmasks and step states
into device tensors — capped at 64 steps per workflow for the bitmask path
(larger workflows stay host-evaluated).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

# step types (models.go:8-25)
STEP_TYPES = {
    "llm", "worker", "http", "container", "script", "approval", "input",
    "condition", "switch", "parallel", "loop", "delay", "notify",
    "transform", "storage", "subworkflow",
}

# run status (models.go:28-38)
RUN_PENDING = "pending"
RUN_RUNNING = "running"
RUN_WAITING = "waiting"
RUN_SUCCEEDED = "succeeded"
RUN_FAILED = "failed"
RUN_CANCELLED = "cancelled"
RUN_TIMED_OUT = "timed_out"
RUN_TERMINAL = {RUN_SUCCEEDED, RUN_FAILED, RUN_CANCELLED, RUN_TIMED_OUT}

# step status (models.go:41-52)
STEP_PENDING = "pending"
STEP_RUNNING = "running"
STEP_WAITING = "waiting"
STEP_SUCCEEDED = "succeeded"
STEP_FAILED = "failed"
STEP_CANCELLED = "cancelled"
STEP_TIMED_OUT = "timed_out"
STEP_TERMINAL = {STEP_SUCCEEDED, STEP_FAILED, STEP_CANCELLED, STEP_TIMED_OUT}


@dataclass
class RetryConfig:
    max_retries: int = 0
    initial_backoff_sec: int = 0
    max_backoff_sec: int = 0
    multiplier: float = 0.0

    @classmethod
    def from_dict(cls, d):
        if not d:
            return None
        return cls(
            max_retries=int(d.get("max_retries", 0) or 0),
            initial_backoff_sec=int(d.get("initial_backoff_sec", 0) or 0),
            max_backoff_sec=int(d.get("max_backoff_sec", 0) or 0),
            multiplier=float(d.get("multiplier", 0) or 0),
        )

    def to_dict(self):
        return {
            "max_retries": self.max_retries,
            "initial_backoff_sec": self.initial_backoff_sec,
            "max_backoff_sec": self.max_backoff_sec,
            "multiplier": self.multiplier,
        }


@dataclass
class StepMeta:
    actor_id: str = ""
    actor_type: str = ""
    idempotency_key: str = ""
    pack_id: str = ""
    capability: str = ""
    risk_tags: List[str] = field(default_factory=list)
    requires: List[str] = field(default_factory=list)
    labels: Dict[str, str] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, d):
        if not d:
            return None
        return cls(
            actor_id=str(d.get("actor_id", "") or ""),
            actor_type=str(d.get("actor_type", "") or ""),
            idempotency_key=str(d.get("idempotency_key", "") or ""),
            pack_id=str(d.get("pack_id", "") or ""),
            capability=str(d.get("capability", "") or ""),
            risk_tags=list(d.get("risk_tags", []) or []),
            requires=list(d.get("requires", []) or []),
            labels=dict(d.get("labels", {}) or {}),
        )


@dataclass
class Step:
    id: str = ""
    name: str = ""
    type: str = "worker"
    worker_id: str = ""
    topic: str = ""
    depends_on: List[str] = field(default_factory=list)
    condition: str = ""
    for_each: str = ""
    max_parallel: int = 0
    input: Dict[str, Any] = field(default_factory=dict)
    input_schema: Dict[str, Any] = field(default_factory=dict)
    input_schema_id: str = ""
    output_path: str = ""
    output_schema: Dict[str, Any] = field(default_factory=dict)
    output_schema_id: str = ""
    meta: Optional[StepMeta] = None
    on_error: str = ""
    retry: Optional[RetryConfig] = None
    timeout_sec: int = 0
    delay_sec: int = 0
    delay_until: str = ""
    route_labels: Dict[str, str] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, sid: str, d: Dict[str, Any]) -> "Step":
        return cls(
            id=str(d.get("id", sid) or sid),
            name=str(d.get("name", "") or ""),
            type=str(d.get("type", "worker") or "worker"),
            worker_id=str(d.get("worker_id", "") or ""),
            topic=str(d.get("topic", "") or ""),
            depends_on=list(d.get("depends_on", []) or []),
            condition=str(d.get("condition", "") or ""),
            for_each=str(d.get("for_each", "") or ""),
            max_parallel=int(d.get("max_parallel", 0) or 0),
            input=dict(d.get("input", {}) or {}),
            input_schema=dict(d.get("input_schema", {}) or {}),
            input_schema_id=str(d.get("input_schema_id", "") or ""),
            output_path=str(d.get("output_path", "") or ""),
            output_schema=dict(d.get("output_schema", {}) or {}),
            output_schema_id=str(d.get("output_schema_id", "") or ""),
            meta=StepMeta.from_dict(d.get("meta")),
            on_error=str(d.get("on_error", "") or ""),
            retry=RetryConfig.from_dict(d.get("retry")),
            timeout_sec=int(d.get("timeout_sec", 0) or 0),
            delay_sec=int(d.get("delay_sec", 0) or 0),
            delay_until=str(d.get("delay_until", "") or ""),
            route_labels=dict(d.get("route_labels", {}) or {}),
        )

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"id": self.id, "type": self.type}
        for k in ("name", "worker_id", "topic", "condition", "for_each", "output_path",
                  "input_schema_id", "output_schema_id", "on_error", "delay_until"):
            v = getattr(self, k)
            if v:
                d[k] = v
        for k in ("max_parallel", "timeout_sec", "delay_sec"):
            v = getattr(self, k)
            if v:
                d[k] = v
        for k in ("depends_on", "input", "input_schema", "output_schema", "route_labels"):
            v = getattr(self, k)
            if v:
                d[k] = v
        if self.retry is not None:
            d["retry"] = self.retry.to_dict()
        if self.meta is not None:
            d["meta"] = {k: v for k, v in vars(self.meta).items() if v}
        return d


@dataclass
class Workflow:
    id: str = ""
    org_id: str = ""
    team_id: str = ""
    name: str = ""
    description: str = ""
    version: str = ""
    timeout_sec: int = 0
    steps: Dict[str, Step] = field(default_factory=dict)
    config: Dict[str, Any] = field(default_factory=dict)
    input_schema: Dict[str, Any] = field(default_factory=dict)
    parameters: List[Dict[str, Any]] = field(default_factory=list)
    created_by: str = ""
    created_at: float = 0.0
    updated_at: float = 0.0

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Workflow":
        steps = {}
        for sid, sd in (d.get("steps", {}) or {}).items():
            steps[sid] = Step.from_dict(sid, sd or {})
        return cls(
            id=str(d.get("id", "") or ""),
            org_id=str(d.get("org_id", "") or ""),
            team_id=str(d.get("team_id", "") or ""),
            name=str(d.get("name", "") or ""),
            description=str(d.get("description", "") or ""),
            version=str(d.get("version", "") or ""),
            timeout_sec=int(d.get("timeout_sec", 0) or 0),
            steps=steps,
            config=dict(d.get("config", {}) or {}),
            input_schema=dict(d.get("input_schema", {}) or {}),
            parameters=list(d.get("parameters", []) or []),
            created_by=str(d.get("created_by", "") or ""),
            created_at=float(d.get("created_at", 0) or 0),
            updated_at=float(d.get("updated_at", 0) or 0),
        )

    def to_dict(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "org_id": self.org_id,
            "team_id": self.team_id,
            "name": self.name,
            "description": self.description,
            "version": self.version,
            "timeout_sec": self.timeout_sec,
            "steps": {sid: s.to_dict() for sid, s in self.steps.items()},
            "config": self.config,
            "input_schema": self.input_schema,
            "parameters": self.parameters,
            "created_by": self.created_by,
            "created_at": self.created_at,
            "updated_at": self.updated_at,
        }


@dataclass
class StepRun:
    step_id: str = ""
    status: str = ""  # "" = not started
    started_at: Optional[float] = None
    completed_at: Optional[float] = None
    next_attempt_at: Optional[float] = None
    attempts: int = 0
    input: Optional[Dict[str, Any]] = None
    output: Any = None
    error: Optional[Dict[str, Any]] = None
    job_id: str = ""
    item: Any = None
    children: Dict[str, "StepRun"] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"step_id": self.step_id, "status": self.status}
        for k in ("started_at", "completed_at", "next_attempt_at", "input", "output", "error", "item"):
            v = getattr(self, k)
            if v is not None:
                d[k] = v
        if self.attempts:
            d["attempts"] = self.attempts
        if self.job_id:
            d["job_id"] = self.job_id
        if self.children:
            d["children"] = {cid: c.to_dict() for cid, c in self.children.items()}
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "StepRun":
        sr = cls(
            step_id=str(d.get("step_id", "") or ""),
            status=str(d.get("status", "") or ""),
            started_at=d.get("started_at"),
            completed_at=d.get("completed_at"),
            next_attempt_at=d.get("next_attempt_at"),
            attempts=int(d.get("attempts", 0) or 0),
            input=d.get("input"),
            output=d.get("output"),
            error=d.get("error"),
            job_id=str(d.get("job_id", "") or ""),
            item=d.get("item"),
        )
        for cid, cd in (d.get("children", {}) or {}).items():
            sr.children[cid] = StepRun.from_dict(cd)
        return sr

    def clone(self) -> "StepRun":
        return StepRun.from_dict(self.to_dict())


@dataclass
class WorkflowRun:
    id: str = ""
    workflow_id: str = ""
    org_id: str = ""
    team_id: str = ""
    input: Dict[str, Any] = field(default_factory=dict)
    context: Dict[str, Any] = field(default_factory=dict)
    status: str = RUN_PENDING
    started_at: Optional[float] = None
    completed_at: Optional[float] = None
    output: Dict[str, Any] = field(default_factory=dict)
    error: Dict[str, Any] = field(default_factory=dict)
    steps: Dict[str, StepRun] = field(default_factory=dict)
    triggered_by: str = ""
    created_at: float = 0.0
    updated_at: float = 0.0
    labels: Dict[str, str] = field(default_factory=dict)
    metadata: Dict[str, str] = field(default_factory=dict)
    idempotency_key: str = ""
    rerun_of: str = ""
    rerun_step: str = ""
    dry_run: bool = False

    def to_dict(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "workflow_id": self.workflow_id,
            "org_id": self.org_id,
            "team_id": self.team_id,
            "input": self.input,
            "context": self.context,
            "status": self.status,
            "started_at": self.started_at,
            "completed_at": self.completed_at,
            "output": self.output,
            "error": self.error,
            "steps": {sid: sr.to_dict() for sid, sr in self.steps.items()},
            "triggered_by": self.triggered_by,
            "created_at": self.created_at,
            "updated_at": self.updated_at,
            "labels": self.labels,
            "metadata": self.metadata,
            "idempotency_key": self.idempotency_key,
            "rerun_of": self.rerun_of,
            "rerun_step": self.rerun_step,
            "dry_run": self.dry_run,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "WorkflowRun":
        run = cls(
            id=str(d.get("id", "") or ""),
            workflow_id=str(d.get("workflow_id", "") or ""),
            org_id=str(d.get("org_id", "") or ""),
            team_id=str(d.get("team_id", "") or ""),
            input=dict(d.get("input", {}) or {}),
            context=dict(d.get("context", {}) or {}),
            status=str(d.get("status", RUN_PENDING) or RUN_PENDING),
            started_at=d.get("started_at"),
            completed_at=d.get("completed_at"),
            output=dict(d.get("output", {}) or {}),
            error=dict(d.get("error", {}) or {}),
            triggered_by=str(d.get("triggered_by", "") or ""),
            created_at=float(d.get("created_at", 0) or 0),
            updated_at=float(d.get("updated_at", 0) or 0),
            labels=dict(d.get("labels", {}) or {}),
            metadata=dict(d.get("metadata", {}) or {}),
            idempotency_key=str(d.get("idempotency_key", "") or ""),
            rerun_of=str(d.get("rerun_of", "") or ""),
            rerun_step=str(d.get("rerun_step", "") or ""),
            dry_run=bool(d.get("dry_run", False)),
        )
        for sid, sd in (d.get("steps", {}) or {}).items():
            run.steps[sid] = StepRun.from_dict(sd or {})
        return run


@dataclass
class TimelineEvent:
    time: float = 0.0
    type: str = ""
    run_id: str = ""
    workflow_id: str = ""
    step_id: str = ""
    job_id: str = ""
    status: str = ""
    result_ptr: str = ""
    message: str = ""
    data: Optional[Dict[str, Any]] = None

    def to_dict(self) -> Dict[str, Any]:
        d = {"time": self.time, "type": self.type}
        for k in ("run_id", "workflow_id", "step_id", "job_id", "status", "result_ptr", "message"):
            v = getattr(self, k)
            if v:
                d[k] = v
        if self.data:
            d["data"] = self.data
        return d
