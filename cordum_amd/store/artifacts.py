"""Artifact store with retention classes.

Oracle: core/infra/artifacts/redis_store.go:18-161 — Put(content, meta) ->
`redis://art:<uuid>`; retention classes short/standard/audit -> TTL 24h/7d/30d
(env-tunable); metadata at `art:meta:<id>`.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple

from ..utils.clock import Clock, SYSTEM_CLOCK
from ..utils.ids import new_id
from .memory_store import MemoryStore

RETENTION_TTLS = {
    "short": 24 * 3600,
    "standard": 7 * 24 * 3600,
    "audit": 30 * 24 * 3600,
}


@dataclass
class ArtifactMeta:
    id: str
    content_type: str = "application/octet-stream"
    retention: str = "standard"
    size: int = 0
    labels: Dict[str, str] = field(default_factory=dict)
    created_at: int = 0


class ArtifactStore:
    def __init__(self, memory: MemoryStore, clock: Clock = SYSTEM_CLOCK):
        self._memory = memory
        self._clock = clock
        self._meta: Dict[str, ArtifactMeta] = {}

    def put(
        self,
        content: bytes,
        content_type: str = "application/octet-stream",
        retention: str = "standard",
        labels: Optional[Dict[str, str]] = None,
    ) -> str:
        if retention not in RETENTION_TTLS:
            retention = "standard"
        art_id = new_id()
        key = f"art:{art_id}"
        ptr = self._memory.put(key, content, ttl_s=RETENTION_TTLS[retention])
        self._meta[art_id] = ArtifactMeta(
            id=art_id,
            content_type=content_type,
            retention=retention,
            size=len(content),
            labels=dict(labels or {}),
            created_at=self._clock.now_micros(),
        )
        return ptr

    def get(self, art_id: str) -> Optional[Tuple[bytes, ArtifactMeta]]:
        blob = self._memory.get(f"art:{art_id}")
        meta = self._meta.get(art_id)
        if blob is None or meta is None:
            return None
        return blob, meta

    def get_pointer(self, ptr: str) -> Optional[Tuple[bytes, ArtifactMeta]]:
        key = ptr[len("redis://"):] if ptr.startswith("redis://") else ptr
        if not key.startswith("art:"):
            return None
        return self.get(key[len("art:"):])
