"""Reconciler (timeout scans) + pending replayer.

Oracles: scheduler/reconciler.go:12-144 (stale DISPATCHED/RUNNING -> TIMEOUT
via per-state indexes with cutoffs; deadline expirations -> TIMEOUT; leader
lock `cordum:reconciler:default`) and scheduler/pending_replayer.go:12-106
(PENDING jobs older than pendingAge re-driven through handleJobRequest from
the persisted job request).

Tick-driven (no internal goroutine/thread): the single-process runtime calls
`tick()` from its control loop, and the batched GPU path runs the same scans
as the K4 deadline/timeout kernel (ops/hip/state_kernels.hip) over the
device job table.
"""
from __future__ import annotations


from ..protocol import JobState
from ..store import InvalidTransition, JobStore
from ..utils.clock import Clock, SYSTEM_CLOCK

DEFAULT_DISPATCH_TIMEOUT_S = 300.0
DEFAULT_RUNNING_TIMEOUT_S = 9000.0
DEFAULT_POLL_INTERVAL_S = 30.0
DEFAULT_PENDING_AGE_S = 300.0

RECONCILER_LOCK = "cordum:reconciler:default"
REPLAYER_LOCK = "cordum:replayer:pending"


class Reconciler:
    def __init__(
        self,
        store: JobStore,
        dispatch_timeout_s: float = DEFAULT_DISPATCH_TIMEOUT_S,
        running_timeout_s: float = DEFAULT_RUNNING_TIMEOUT_S,
        poll_interval_s: float = DEFAULT_POLL_INTERVAL_S,
        clock: Clock = SYSTEM_CLOCK,
        owner: str = "reconciler-0",
    ):
        self.store = store
        self.dispatch_timeout_s = dispatch_timeout_s
        self.running_timeout_s = running_timeout_s
        self.poll_interval_s = poll_interval_s
        self.clock = clock
        self.owner = owner

    def update_timeouts(self, dispatch_timeout_s: float = 0, running_timeout_s: float = 0) -> None:
        if dispatch_timeout_s > 0:
            self.dispatch_timeout_s = dispatch_timeout_s
        if running_timeout_s > 0:
            self.running_timeout_s = running_timeout_s

    def tick(self) -> int:
        """Run one reconciliation pass (leader-locked); returns #jobs timed out."""
        if not self.store.try_lock(RECONCILER_LOCK, self.owner, ttl_s=self.poll_interval_s * 2):
            return 0
        try:
            n = 0
            now_us = self.clock.now_micros()
            n += self._timeout_state(JobState.DISPATCHED, now_us - int(self.dispatch_timeout_s * 1e6))
            n += self._timeout_state(JobState.RUNNING, now_us - int(self.running_timeout_s * 1e6))
            n += self._expire_deadlines(now_us)
            return n
        finally:
            self.store.unlock(RECONCILER_LOCK, self.owner)

    def _timeout_state(self, state: JobState, cutoff_us: int) -> int:
        n = 0
        for _ in range(100):  # maxIterations
            jobs = self.store.list_jobs_by_state(state, updated_before_micros=cutoff_us, limit=200)
            if not jobs:
                break
            progress = 0
            for job_id in jobs:
                try:
                    self.store.set_state(job_id, JobState.TIMEOUT)
                    progress += 1
                    n += 1
                except (InvalidTransition, ValueError):
                    pass
            if progress == 0:
                break
        return n

    def _expire_deadlines(self, now_us: int) -> int:
        n = 0
        for job_id in self.store.list_expired_deadlines(now_us, limit=200):
            try:
                self.store.set_state(job_id, JobState.TIMEOUT)
                n += 1
            except (InvalidTransition, ValueError):
                self.store.clear_deadline(job_id)
        return n


class PendingReplayer:
    def __init__(
        self,
        engine,
        store: JobStore,
        pending_age_s: float = DEFAULT_PENDING_AGE_S,
        poll_interval_s: float = DEFAULT_POLL_INTERVAL_S,
        clock: Clock = SYSTEM_CLOCK,
        owner: str = "replayer-0",
    ):
        self.engine = engine
        self.store = store
        self.pending_age_s = pending_age_s
        self.poll_interval_s = poll_interval_s
        self.clock = clock
        self.owner = owner

    def tick(self) -> int:
        if not self.store.try_lock(REPLAYER_LOCK, self.owner, ttl_s=self.poll_interval_s * 2):
            return 0
        try:
            cutoff_us = self.clock.now_micros() - int(self.pending_age_s * 1e6)
            jobs = self.store.list_jobs_by_state(JobState.PENDING, updated_before_micros=cutoff_us, limit=200)
            n = 0
            for job_id in jobs:
                req = self.store.get_job_request(job_id)
                if req is None:
                    continue
                trace_id = self.store.get_job_meta(job_id).get("trace_id", "")
                try:
                    self.engine.handle_job_request(req, trace_id)
                    n += 1
                except Exception:
                    pass
            return n
        finally:
            self.store.unlock(REPLAYER_LOCK, self.owner)
