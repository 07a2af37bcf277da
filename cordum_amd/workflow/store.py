"""Workflow + run store.

Oracle: core/workflow/store_redis.go:49-523 — `wf:def:<id>` definitions with
org/all indexes, `wf:run:<id>` runs with per-workflow/all/status indexes and
per-org active set, append-only timeline `wf:run:timeline:<id>` capped at
1000 entries, run idempotency `wf:run:idempotency:<key>` (SETNX).
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional, Tuple

from ..utils.clock import Clock, SYSTEM_CLOCK
from .models import RUN_TERMINAL, TimelineEvent, Workflow, WorkflowRun

TIMELINE_CAP = 1000


class RunNotFound(KeyError):
    pass


class WorkflowNotFound(KeyError):
    pass


class WorkflowStore:
    def __init__(self, clock: Clock = SYSTEM_CLOCK):
        self._clock = clock
        self._mu = threading.RLock()
        self._workflows: Dict[str, Workflow] = {}
        self._runs: Dict[str, WorkflowRun] = {}
        self._timelines: Dict[str, List[TimelineEvent]] = {}
        self._idempotency: Dict[str, str] = {}

    # -- workflows ------------------------------------------------------------
    def put_workflow(self, wf: Workflow) -> None:
        with self._mu:
            now = self._clock.now()
            if wf.id in self._workflows:
                wf.created_at = self._workflows[wf.id].created_at or now
            else:
                wf.created_at = wf.created_at or now
            wf.updated_at = now
            self._workflows[wf.id] = wf

    def get_workflow(self, wf_id: str) -> Workflow:
        with self._mu:
            wf = self._workflows.get(wf_id)
            if wf is None:
                raise WorkflowNotFound(wf_id)
            return wf

    def delete_workflow(self, wf_id: str) -> bool:
        with self._mu:
            return self._workflows.pop(wf_id, None) is not None

    def list_workflows(self, org_id: str = "") -> List[Workflow]:
        with self._mu:
            out = [w for w in self._workflows.values() if not org_id or w.org_id == org_id]
            return sorted(out, key=lambda w: w.id)

    # -- runs -------------------------------------------------------------------
    def create_run(self, run: WorkflowRun) -> None:
        with self._mu:
            run.created_at = run.created_at or self._clock.now()
            run.updated_at = self._clock.now()
            self._runs[run.id] = run

    def get_run(self, run_id: str) -> WorkflowRun:
        with self._mu:
            run = self._runs.get(run_id)
            if run is None:
                raise RunNotFound(run_id)
            return run

    def update_run(self, run: WorkflowRun) -> None:
        with self._mu:
            run.updated_at = self._clock.now()
            self._runs[run.id] = run

    def delete_run(self, run_id: str) -> bool:
        with self._mu:
            self._timelines.pop(run_id, None)
            return self._runs.pop(run_id, None) is not None

    def list_runs(
        self,
        workflow_id: str = "",
        status: str = "",
        org_id: str = "",
        limit: int = 100,
        cursor: Optional[float] = None,
    ) -> Tuple[List[WorkflowRun], Optional[float]]:
        with self._mu:
            runs = [
                r
                for r in self._runs.values()
                if (not workflow_id or r.workflow_id == workflow_id)
                and (not status or r.status == status)
                and (not org_id or r.org_id == org_id)
            ]
            runs.sort(key=lambda r: -r.created_at)
            if cursor is not None:
                runs = [r for r in runs if r.created_at < cursor]
            page = runs[:limit]
            next_cursor = page[-1].created_at if len(runs) > limit and page else None
            return page, next_cursor

    def active_runs(self, org_id: str = "") -> List[WorkflowRun]:
        with self._mu:
            return [
                r
                for r in self._runs.values()
                if r.status not in RUN_TERMINAL and (not org_id or r.org_id == org_id)
            ]

    def count_active_runs(self, org_id: str = "") -> int:
        return len(self.active_runs(org_id))

    # -- timeline ----------------------------------------------------------------
    def append_timeline(self, run_id: str, event: TimelineEvent) -> None:
        with self._mu:
            tl = self._timelines.setdefault(run_id, [])
            tl.append(event)
            if len(tl) > TIMELINE_CAP:
                del tl[: len(tl) - TIMELINE_CAP]

    def get_timeline(self, run_id: str) -> List[TimelineEvent]:
        with self._mu:
            return list(self._timelines.get(run_id, []))

    # -- idempotency --------------------------------------------------------------
    def try_set_run_idempotency(self, key: str, run_id: str) -> Tuple[bool, str]:
        with self._mu:
            existing = self._idempotency.get(key)
            if existing is not None:
                return False, existing
            self._idempotency[key] = run_id
            return True, run_id

    # -- snapshot (WAL) -------------------------------------------------------------
    def snapshot(self) -> Dict:
        with self._mu:
            out: Dict = {}
            for wid, wf in self._workflows.items():
                out[f"wf:def:{wid}"] = wf.to_dict()
            for rid, run in self._runs.items():
                out[f"wf:run:{rid}"] = run.to_dict()
            for rid, tl in self._timelines.items():
                out[f"wf:run:timeline:{rid}"] = [e.to_dict() for e in tl]
            for k, v in self._idempotency.items():
                out[f"wf:run:idempotency:{k}"] = v
            return out

    def restore(self, snap: Dict) -> None:
        with self._mu:
            for key, val in snap.items():
                if key.startswith("wf:def:"):
                    wf = Workflow.from_dict(val)
                    self._workflows[wf.id] = wf
                elif key.startswith("wf:run:timeline:"):
                    rid = key[len("wf:run:timeline:"):]
                    self._timelines[rid] = [TimelineEvent(**e) for e in val]
                elif key.startswith("wf:run:idempotency:"):
                    self._idempotency[key[len("wf:run:idempotency:"):]] = val
                elif key.startswith("wf:run:"):
                    run = WorkflowRun.from_dict(val)
                    self._runs[run.id] = run
