"""Dead-letter queue store.

Oracle: core/infra/memory/dlq_store.go:14-180 — `dlq:entry:<job_id>` JSON
{topic, status, reason, reason_code, last_state, attempts}, index capped
~1000, list/page/get/delete.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, asdict
from typing import Dict, List, Optional, Tuple

from ..utils.clock import Clock, SYSTEM_CLOCK

DLQ_CAP = 1000


@dataclass
class DLQEntry:
    job_id: str
    topic: str = ""
    status: str = ""
    reason: str = ""
    reason_code: str = ""
    last_state: str = ""
    attempts: int = 0
    tenant: str = ""
    trace_id: str = ""
    created_at: int = 0  # micros

    def to_dict(self) -> Dict:
        return asdict(self)


class DLQStore:
    def __init__(self, clock: Clock = SYSTEM_CLOCK, cap: int = DLQ_CAP):
        self._clock = clock
        self._cap = cap
        self._mu = threading.Lock()
        self._entries: Dict[str, DLQEntry] = {}
        self._order: Dict[str, int] = {}  # job_id -> score (micros)

    def add(self, entry: DLQEntry) -> None:
        with self._mu:
            if not entry.created_at:
                entry.created_at = self._clock.now_micros()
            self._entries[entry.job_id] = entry
            self._order[entry.job_id] = entry.created_at
            if len(self._order) > self._cap:
                for victim in sorted(self._order, key=self._order.get)[: len(self._order) - self._cap]:
                    self._order.pop(victim, None)
                    self._entries.pop(victim, None)

    def get(self, job_id: str) -> Optional[DLQEntry]:
        with self._mu:
            return self._entries.get(job_id)

    def delete(self, job_id: str) -> bool:
        with self._mu:
            self._order.pop(job_id, None)
            return self._entries.pop(job_id, None) is not None

    def list(self, limit: int = 100, cursor: Optional[int] = None) -> Tuple[List[DLQEntry], Optional[int]]:
        with self._mu:
            items = sorted(self._entries.values(), key=lambda e: -e.created_at)
            if cursor is not None:
                items = [e for e in items if e.created_at < cursor]
            page = items[:limit]
            next_cursor = page[-1].created_at if len(items) > limit and page else None
            return page, next_cursor

    def __len__(self) -> int:
        with self._mu:
            return len(self._entries)

    def snapshot(self) -> Dict:
        with self._mu:
            return {f"dlq:entry:{j}": e.to_dict() for j, e in self._entries.items()}

    def restore(self, snap: Dict) -> None:
        with self._mu:
            for key, d in snap.items():
                if key.startswith("dlq:entry:"):
                    e = DLQEntry(**d)
                    self._entries[e.job_id] = e
                    self._order[e.job_id] = e.created_at
