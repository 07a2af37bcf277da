"""Host job store — the durable/API view of the job table.

Semantics oracle: core/infra/memory/job_store.go (state machine with
allowed-transition enforcement :249-329, per-state ZSET indexes, `job:recent`
capped at 1000, deadline index :607-675, event log `ts|state` :319, protojson
job requests :575-605, trace sets :800-815, tenant active sets, scoped
idempotency :924, safety decision :1036 and approval records :1108,
per-job locks :203-217).

Key naming is kept Redis-compatible (`job:meta:<id>`, `job:index:<STATE>` …)
so the WAL/checkpoint layout (store/wal.py) matches the reference's Redis
key scheme. On the GPU data plane the HOT copy of {state, deadline, attempts,
tenant, topic} lives in the HBM job table (ops/pipeline.py); this host store
is the system of record for everything the API can read back.
"""
from __future__ import annotations

import threading
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

from ..protocol import JobState, can_transition, is_terminal, parse_state
from ..protocol.capv2 import JobRequest, PolicyConstraints, PolicyRemediation
from ..utils.clock import Clock, SYSTEM_CLOCK

ACTIVE_STATES = frozenset(
    {
        JobState.PENDING,
        JobState.APPROVAL_REQUIRED,
        JobState.SCHEDULED,
        JobState.DISPATCHED,
        JobState.RUNNING,
    }
)

RECENT_CAP = 1000
DEFAULT_META_TTL_S = 7 * 24 * 3600  # JOB_META_TTL default (job_store.go:85)


class InvalidTransition(Exception):
    def __init__(self, frm: JobState, to: JobState):
        super().__init__(f"invalid transition {frm} -> {to}")
        self.frm = frm
        self.to = to


@dataclass
class SafetyDecisionRecord:
    """Scheduler-side decision record (scheduler/safety_client.go:104-117)."""

    decision: str = ""  # allow|deny|require_approval|allow_with_constraints|throttle
    reason: str = ""
    rule_id: str = ""
    policy_snapshot: str = ""
    constraints: Optional[PolicyConstraints] = None
    approval_required: bool = False
    approval_ref: str = ""
    remediations: List[PolicyRemediation] = field(default_factory=list)
    job_hash: str = ""
    checked_at: int = 0  # micros


@dataclass
class ApprovalRecord:
    approved_by: str = ""
    role: str = ""
    approved_at: int = 0  # micros
    reason: str = ""
    note: str = ""
    policy_snapshot: str = ""
    job_hash: str = ""
    decision: str = ""  # "approved"|"rejected"


@dataclass
class _JobEntry:
    meta: Dict[str, Any] = field(default_factory=dict)
    events: List[str] = field(default_factory=list)
    request: Optional[JobRequest] = None
    safety: Optional[SafetyDecisionRecord] = None
    approval: Optional[ApprovalRecord] = None


class JobStore:
    """In-process job store with the reference's Redis-store semantics."""

    def __init__(self, clock: Clock = SYSTEM_CLOCK, meta_ttl_s: int = DEFAULT_META_TTL_S):
        self._clock = clock
        self._meta_ttl_s = meta_ttl_s
        self._mu = threading.RLock()
        self._jobs: Dict[str, _JobEntry] = {}
        self._state_index: Dict[JobState, Dict[str, int]] = {s: {} for s in JobState}
        self._recent: Dict[str, int] = {}
        self._deadlines: Dict[str, int] = {}  # job_id -> unix micros
        self._traces: Dict[str, List[str]] = {}
        self._tenant_active: Dict[str, set] = {}
        self._idempotency: Dict[str, str] = {}  # "tenant\x00key" -> job_id
        self._locks: Dict[str, Tuple[str, float]] = {}  # resource -> (owner, expiry)

    # -- time helpers --------------------------------------------------------
    def _now_us(self) -> int:
        return self._clock.now_micros()

    # -- state machine -------------------------------------------------------
    def set_state(self, job_id: str, state: JobState) -> None:
        """job_store.go:249-329 (WATCH/tx equivalent under one lock)."""
        if not job_id or state == JobState.UNSPECIFIED:
            raise ValueError("invalid jobID or state")
        now = self._now_us()
        with self._mu:
            entry = self._jobs.setdefault(job_id, _JobEntry())
            prev = parse_state(entry.meta.get("state", ""))
            if not can_transition(prev, state):
                raise InvalidTransition(prev, state)
            attempts = int(entry.meta.get("attempts", 0) or 0)
            if state == JobState.SCHEDULED and prev != JobState.SCHEDULED:
                attempts += 1
            entry.meta.update(state=state.name, updated_at=now, attempts=attempts)
            self._state_index[prev].pop(job_id, None)
            self._state_index[state][job_id] = now
            tenant = entry.meta.get("tenant", "")
            if tenant:
                active = self._tenant_active.setdefault(tenant, set())
                if state in ACTIVE_STATES:
                    active.add(job_id)
                elif is_terminal(state):
                    active.discard(job_id)
            # move-to-end keeps insertion order == recency order, so the
            # cap evicts from the front in O(1) (the previous full sort per
            # transition was ~40% of the host submit path at scale)
            self._recent.pop(job_id, None)
            self._recent[job_id] = now
            while len(self._recent) > RECENT_CAP:
                del self._recent[next(iter(self._recent))]
            entry.events.append(f"{now}|{state.name}")
            if is_terminal(state):
                self._deadlines.pop(job_id, None)
                entry.meta.pop("deadline_unix", None)

    def get_state(self, job_id: str) -> JobState:
        with self._mu:
            entry = self._jobs.get(job_id)
            if entry is None:
                return JobState.UNSPECIFIED
            return parse_state(entry.meta.get("state", ""))

    def cancel_job(self, job_id: str) -> bool:
        """CancelJob (job_store.go:137-201): non-terminal -> CANCELLED."""
        with self._mu:
            st = self.get_state(job_id)
            if is_terminal(st):
                return False
            if st == JobState.UNSPECIFIED:
                return False
            if st in (JobState.PENDING, JobState.APPROVAL_REQUIRED):
                # reference routes PENDING/APPROVAL through FAILED-able paths;
                # cancellation of not-yet-scheduled jobs marks meta directly.
                entry = self._jobs.setdefault(job_id, _JobEntry())
                now = self._now_us()
                prev = parse_state(entry.meta.get("state", ""))
                entry.meta.update(state=JobState.CANCELLED.name, updated_at=now)
                self._state_index[prev].pop(job_id, None)
                self._state_index[JobState.CANCELLED][job_id] = now
                entry.events.append(f"{now}|{JobState.CANCELLED.name}")
                tenant = entry.meta.get("tenant", "")
                if tenant:
                    self._tenant_active.setdefault(tenant, set()).discard(job_id)
                self._deadlines.pop(job_id, None)
                return True
            self.set_state(job_id, JobState.CANCELLED)
            return True

    # -- meta ---------------------------------------------------------------
    def set_job_meta(self, job_id: str, **fields: Any) -> None:
        with self._mu:
            entry = self._jobs.setdefault(job_id, _JobEntry())
            entry.meta.update({k: v for k, v in fields.items() if v is not None})
            entry.meta["updated_at"] = self._now_us()

    def get_job_meta(self, job_id: str) -> Dict[str, Any]:
        with self._mu:
            entry = self._jobs.get(job_id)
            return dict(entry.meta) if entry else {}

    def set_topic(self, job_id: str, topic: str) -> None:
        self.set_job_meta(job_id, topic=topic)

    def set_tenant(self, job_id: str, tenant: str) -> None:
        self.set_job_meta(job_id, tenant=tenant)

    def set_trace(self, job_id: str, trace_id: str) -> None:
        self.set_job_meta(job_id, trace_id=trace_id)

    def set_result_ptr(self, job_id: str, ptr: str) -> None:
        self.set_job_meta(job_id, result_ptr=ptr)

    def set_worker(self, job_id: str, worker_id: str) -> None:
        self.set_job_meta(job_id, worker_id=worker_id)

    def set_error(self, job_id: str, code: str, message: str) -> None:
        self.set_job_meta(job_id, error_code=code, error_message=message)

    # -- request persistence (job:req:<id>, job_store.go:575-605) ------------
    def set_job_request(self, job_id: str, req: JobRequest) -> None:
        with self._mu:
            self._jobs.setdefault(job_id, _JobEntry()).request = req.copy()

    def get_job_request(self, job_id: str) -> Optional[JobRequest]:
        with self._mu:
            entry = self._jobs.get(job_id)
            return entry.request.copy() if entry and entry.request else None

    # -- deadlines (job:deadline ZSET, job_store.go:607-675) ------------------
    def set_deadline(self, job_id: str, at_unix_micros: int) -> None:
        with self._mu:
            self._deadlines[job_id] = at_unix_micros
            self._jobs.setdefault(job_id, _JobEntry()).meta["deadline_unix"] = at_unix_micros

    def clear_deadline(self, job_id: str) -> None:
        with self._mu:
            self._deadlines.pop(job_id, None)

    def list_expired_deadlines(self, now_unix_micros: Optional[int] = None, limit: int = 1000) -> List[str]:
        cutoff = now_unix_micros if now_unix_micros is not None else self._now_us()
        with self._mu:
            due = [j for j, at in self._deadlines.items() if at <= cutoff]
            due.sort(key=lambda j: self._deadlines[j])
            return due[:limit]

    # -- per-state index scans (reconciler) -----------------------------------
    def list_jobs_by_state(
        self, state: JobState, updated_before_micros: Optional[int] = None, limit: int = 1000
    ) -> List[str]:
        with self._mu:
            idx = self._state_index[state]
            items = (
                [(j, t) for j, t in idx.items() if t <= updated_before_micros]
                if updated_before_micros is not None
                else list(idx.items())
            )
            items.sort(key=lambda kv: kv[1])
            return [j for j, _ in items[:limit]]

    def count_jobs_by_state(self, state: JobState) -> int:
        with self._mu:
            return len(self._state_index[state])

    def list_jobs_by_state_page(
        self, state: JobState, cursor: Optional[int] = None, limit: int = 50
    ) -> Tuple[List[str], Optional[int]]:
        """Newest-first page over the per-state ZSET analog with a micros
        cursor (job_store.go:676+ / gateway.go:918-1009): state-filtered
        listings page the state index directly instead of post-filtering the
        recent index, so results stay complete at any scale."""
        with self._mu:
            items = sorted(self._state_index[state].items(), key=lambda kv: -kv[1])
            if cursor is not None:
                items = [(j, t) for j, t in items if t < cursor]
            page = items[:limit]
            next_cursor = page[-1][1] if len(items) > limit and page else None
            return [j for j, _ in page], next_cursor

    def list_recent(self, limit: int = 100, cursor: Optional[int] = None) -> Tuple[List[str], Optional[int]]:
        """Recent jobs newest-first with micros cursor (gateway.go:918-1009)."""
        with self._mu:
            items = sorted(self._recent.items(), key=lambda kv: -kv[1])
            if cursor is not None:
                items = [(j, t) for j, t in items if t < cursor]
            page = items[:limit]
            next_cursor = page[-1][1] if len(items) > limit and page else None
            return [j for j, _ in page], next_cursor

    # -- events / traces ------------------------------------------------------
    def get_events(self, job_id: str) -> List[str]:
        with self._mu:
            entry = self._jobs.get(job_id)
            return list(entry.events) if entry else []

    def add_job_to_trace(self, trace_id: str, job_id: str) -> None:
        with self._mu:
            jobs = self._traces.setdefault(trace_id, [])
            if job_id not in jobs:
                jobs.append(job_id)

    def get_trace(self, trace_id: str) -> List[str]:
        with self._mu:
            return list(self._traces.get(trace_id, []))

    # -- tenant concurrency ----------------------------------------------------
    def tenant_active_count(self, tenant: str) -> int:
        with self._mu:
            return len(self._tenant_active.get(tenant, ()))

    # -- idempotency (job:idemp:<tenant>:<key>, job_store.go:924) --------------
    def try_set_idempotency_key(self, tenant: str, key: str, job_id: str) -> Tuple[bool, str]:
        """Returns (inserted, existing_or_new_job_id)."""
        k = f"{tenant}\x00{key}"
        with self._mu:
            existing = self._idempotency.get(k)
            if existing is not None:
                return False, existing
            self._idempotency[k] = job_id
            return True, job_id

    # -- safety / approval records ---------------------------------------------
    def set_safety_decision(self, job_id: str, rec: SafetyDecisionRecord) -> None:
        with self._mu:
            entry = self._jobs.setdefault(job_id, _JobEntry())
            entry.safety = rec
            entry.meta.update(
                safety_decision=rec.decision,
                safety_reason=rec.reason,
                safety_rule_id=rec.rule_id,
                safety_snapshot=rec.policy_snapshot,
                safety_approval_required="true" if rec.approval_required else "false",
                safety_approval_ref=rec.approval_ref,
                safety_job_hash=rec.job_hash,
                safety_checked_at=rec.checked_at or self._now_us(),
            )

    def get_safety_decision(self, job_id: str) -> Optional[SafetyDecisionRecord]:
        with self._mu:
            entry = self._jobs.get(job_id)
            return entry.safety if entry else None

    def set_approval_record(self, job_id: str, rec: ApprovalRecord) -> None:
        with self._mu:
            self._jobs.setdefault(job_id, _JobEntry()).approval = rec

    def get_approval_record(self, job_id: str) -> Optional[ApprovalRecord]:
        with self._mu:
            entry = self._jobs.get(job_id)
            return entry.approval if entry else None

    # -- per-job lock (SETNX+TTL, job_store.go:203-217) -------------------------
    def try_lock(self, resource: str, owner: str, ttl_s: float = 30.0) -> bool:
        now = self._clock.now()
        with self._mu:
            cur = self._locks.get(resource)
            if cur is not None and cur[1] > now and cur[0] != owner:
                return False
            self._locks[resource] = (owner, now + ttl_s)
            return True

    def unlock(self, resource: str, owner: str) -> None:
        with self._mu:
            cur = self._locks.get(resource)
            if cur is not None and cur[0] == owner:
                del self._locks[resource]

    @contextmanager
    def job_lock(self, job_id: str, owner: str = "local", ttl_s: float = 30.0):
        resource = f"lock:job:{job_id}"
        acquired = self.try_lock(resource, owner, ttl_s)
        try:
            yield acquired
        finally:
            if acquired:
                self.unlock(resource, owner)

    # -- snapshot (WAL / checkpoint, Redis-compatible key naming) ---------------
    def snapshot(self) -> Dict[str, Any]:
        with self._mu:
            out: Dict[str, Any] = {}
            for job_id, e in self._jobs.items():
                out[f"job:meta:{job_id}"] = dict(e.meta)
                if e.events:
                    out[f"job:events:{job_id}"] = list(e.events)
                if e.request is not None:
                    out[f"job:req:{job_id}"] = e.request.to_dict()
            out["job:recent"] = dict(self._recent)
            out["job:deadline"] = dict(self._deadlines)
            for t, jobs in self._traces.items():
                out[f"trace:{t}"] = list(jobs)
            for k, v in self._idempotency.items():
                tenant, key = k.split("\x00", 1)
                out[f"job:idemp:{tenant}:{key}"] = v
            return out

    def restore(self, snap: Dict[str, Any]) -> None:
        with self._mu:
            for key, val in snap.items():
                if key.startswith("job:meta:"):
                    job_id = key[len("job:meta:"):]
                    entry = self._jobs.setdefault(job_id, _JobEntry())
                    entry.meta = dict(val)
                    st = parse_state(val.get("state", ""))
                    if st != JobState.UNSPECIFIED:
                        self._state_index[st][job_id] = int(val.get("updated_at", 0))
                    tenant = val.get("tenant", "")
                    if tenant and st in ACTIVE_STATES:
                        self._tenant_active.setdefault(tenant, set()).add(job_id)
                elif key.startswith("job:events:"):
                    self._jobs.setdefault(key[len("job:events:"):], _JobEntry()).events = list(val)
                elif key.startswith("job:req:"):
                    self._jobs.setdefault(key[len("job:req:"):], _JobEntry()).request = JobRequest.from_dict(val)
                elif key == "job:recent":
                    self._recent = dict(val)
                elif key == "job:deadline":
                    self._deadlines = {k: int(v) for k, v in val.items()}
                elif key.startswith("trace:"):
                    self._traces[key[len("trace:"):]] = list(val)
                elif key.startswith("job:idemp:"):
                    rest = key[len("job:idemp:"):]
                    tenant, k2 = rest.split(":", 1)
                    self._idempotency[f"{tenant}\x00{k2}"] = val
