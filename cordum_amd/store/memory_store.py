"""Payload store: context/result/artifact blobs behind opaque pointers.

Oracle: core/infra/memory/redis_store.go:26-160 — `ctx:<job_id>` /
`res:<job_id>` / `art:<id>` values, pointer scheme `redis://<key>`
(PointerForKey/KeyFromPointer :141-160), TTL default 24h (REDIS_DATA_TTL).

The wire pointer format stays `redis://<key>` for client compatibility even
though the backing is a host arena (and, on the data plane, an HBM payload
arena — the rotating payload arenas in ops/pipeline.py and
ops/wf_pipeline.py — with the same keys).
"""
from __future__ import annotations

import threading
from typing import Dict, Optional, Tuple

from ..utils.clock import Clock, SYSTEM_CLOCK

POINTER_SCHEME = "redis://"
DEFAULT_DATA_TTL_S = 24 * 3600


def pointer_for_key(key: str) -> str:
    return POINTER_SCHEME + key


def key_from_pointer(ptr: str) -> str:
    if not ptr.startswith(POINTER_SCHEME):
        raise ValueError(f"invalid pointer {ptr!r}")
    return ptr[len(POINTER_SCHEME):]


class MemoryStore:
    def __init__(self, clock: Clock = SYSTEM_CLOCK, ttl_s: int = DEFAULT_DATA_TTL_S):
        self._clock = clock
        self._ttl_s = ttl_s
        self._mu = threading.Lock()
        self._data: Dict[str, Tuple[bytes, float]] = {}  # key -> (blob, expiry)

    def put(self, key: str, blob: bytes, ttl_s: Optional[int] = None) -> str:
        exp = self._clock.now() + (ttl_s if ttl_s is not None else self._ttl_s)
        with self._mu:
            self._data[key] = (bytes(blob), exp)
        return pointer_for_key(key)

    def put_context(self, job_id: str, blob: bytes) -> str:
        return self.put(f"ctx:{job_id}", blob)

    def put_result(self, job_id: str, blob: bytes) -> str:
        return self.put(f"res:{job_id}", blob)

    def get(self, key: str) -> Optional[bytes]:
        with self._mu:
            ent = self._data.get(key)
            if ent is None:
                return None
            blob, exp = ent
            if exp < self._clock.now():
                del self._data[key]
                return None
            return blob

    def get_pointer(self, ptr: str) -> Optional[bytes]:
        return self.get(key_from_pointer(ptr))

    def delete(self, key: str) -> None:
        with self._mu:
            self._data.pop(key, None)

    def sweep(self) -> int:
        now = self._clock.now()
        with self._mu:
            dead = [k for k, (_, exp) in self._data.items() if exp < now]
            for k in dead:
                del self._data[k]
            return len(dead)

    def keys(self):
        with self._mu:
            return list(self._data.keys())

    def snapshot(self) -> Dict[str, Tuple[bytes, float]]:
        with self._mu:
            return dict(self._data)

    def restore(self, snap) -> None:
        with self._mu:
            self._data = {k: (bytes(b), float(e)) for k, (b, e) in snap.items()}
