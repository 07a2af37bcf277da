"""Resource lock service: shared/exclusive locks with reentrant owner counts.

Oracle: core/infra/locks/redis_store.go:48-279 — three Lua scripts (acquire
with shared→exclusive upgrade when sole owner, release with count decrement,
renew); key `lock:<resource>`; TTL-based expiry. Implemented natively under
one mutex (single process replaces distributed Redis), preserving observable
behavior including upgrade rules.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.clock import Clock, SYSTEM_CLOCK

MODE_SHARED = "shared"
MODE_EXCLUSIVE = "exclusive"


@dataclass
class LockInfo:
    resource: str
    mode: str
    owners: Dict[str, int] = field(default_factory=dict)  # owner -> reentrant count
    expires_at: float = 0.0


class LockService:
    def __init__(self, clock: Clock = SYSTEM_CLOCK):
        self._clock = clock
        self._mu = threading.Lock()
        self._locks: Dict[str, LockInfo] = {}

    def _gc(self, resource: str) -> Optional[LockInfo]:
        info = self._locks.get(resource)
        if info is not None and info.expires_at <= self._clock.now():
            del self._locks[resource]
            return None
        return info

    def acquire(self, resource: str, owner: str, mode: str = MODE_EXCLUSIVE, ttl_s: float = 30.0) -> bool:
        if mode not in (MODE_SHARED, MODE_EXCLUSIVE):
            raise ValueError(f"bad lock mode {mode}")
        now = self._clock.now()
        with self._mu:
            info = self._gc(resource)
            if info is None:
                self._locks[resource] = LockInfo(resource, mode, {owner: 1}, now + ttl_s)
                return True
            if info.mode == MODE_SHARED and mode == MODE_SHARED:
                info.owners[owner] = info.owners.get(owner, 0) + 1
                info.expires_at = max(info.expires_at, now + ttl_s)
                return True
            if owner in info.owners and len(info.owners) == 1:
                # reentrant; shared->exclusive upgrade allowed when sole owner
                info.mode = mode if mode == MODE_EXCLUSIVE else info.mode
                info.owners[owner] += 1
                info.expires_at = max(info.expires_at, now + ttl_s)
                return True
            if info.mode == MODE_EXCLUSIVE and owner in info.owners and mode == MODE_EXCLUSIVE:
                info.owners[owner] += 1
                info.expires_at = max(info.expires_at, now + ttl_s)
                return True
            return False

    def release(self, resource: str, owner: str) -> bool:
        with self._mu:
            info = self._gc(resource)
            if info is None or owner not in info.owners:
                return False
            info.owners[owner] -= 1
            if info.owners[owner] <= 0:
                del info.owners[owner]
            if not info.owners:
                del self._locks[resource]
            return True

    def renew(self, resource: str, owner: str, ttl_s: float = 30.0) -> bool:
        with self._mu:
            info = self._gc(resource)
            if info is None or owner not in info.owners:
                return False
            info.expires_at = self._clock.now() + ttl_s
            return True

    def list(self) -> List[LockInfo]:
        with self._mu:
            for r in list(self._locks):
                self._gc(r)
            return list(self._locks.values())

    def get(self, resource: str) -> Optional[LockInfo]:
        with self._mu:
            return self._gc(resource)
