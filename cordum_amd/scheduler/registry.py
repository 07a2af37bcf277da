"""Worker registry: heartbeat map with TTL expiry + cluster snapshot.

Oracle: scheduler/registry_memory.go:11-113 (30s TTL) and
infra/registry/snapshot.go:9-96 (pool/topic aggregation written to
`sys:workers:snapshot` every 5s).

On the data plane the same information lives in the device worker table
(ops/worker_table.py): per-rank heartbeat rows all-gathered over RCCL each
tick. This host registry is the source the device table is packed from, and
the seam unit tests drive.
"""
from __future__ import annotations

import threading
from typing import Dict

from ..protocol.capv2 import Heartbeat
from ..utils.clock import Clock, SYSTEM_CLOCK

DEFAULT_TTL_S = 30.0


class WorkerRegistry:
    def __init__(self, clock: Clock = SYSTEM_CLOCK, ttl_s: float = DEFAULT_TTL_S):
        self._clock = clock
        self._ttl = ttl_s
        self._mu = threading.Lock()
        self._workers: Dict[str, Heartbeat] = {}
        self._seen: Dict[str, float] = {}

    def update(self, hb: Heartbeat) -> None:
        if not hb.worker_id:
            return
        with self._mu:
            self._workers[hb.worker_id] = hb
            self._seen[hb.worker_id] = self._clock.now()

    def remove(self, worker_id: str) -> None:
        with self._mu:
            self._workers.pop(worker_id, None)
            self._seen.pop(worker_id, None)

    def snapshot(self) -> Dict[str, Heartbeat]:
        """Live workers only (TTL-expired entries dropped)."""
        cutoff = self._clock.now() - self._ttl
        with self._mu:
            dead = [w for w, t in self._seen.items() if t < cutoff]
            for w in dead:
                del self._seen[w]
                del self._workers[w]
            return dict(self._workers)

    def count(self) -> int:
        return len(self.snapshot())

    def cluster_snapshot(self) -> Dict:
        """infra/registry/snapshot.go aggregation: pools/topics/workers."""
        workers = self.snapshot()
        pools: Dict[str, Dict] = {}
        for hb in workers.values():
            pool = hb.pool or "default"
            p = pools.setdefault(pool, {"workers": 0, "active": 0, "capacity": 0})
            p["workers"] += 1
            p["active"] += hb.active_jobs
            p["capacity"] += hb.max_parallel_jobs
        return {
            "pools": pools,
            "workers": [
                {
                    "worker_id": hb.worker_id,
                    "pool": hb.pool,
                    "active_jobs": hb.active_jobs,
                    "max_parallel_jobs": hb.max_parallel_jobs,
                    "cpu_load": hb.cpu_load,
                    "gpu_utilization": hb.gpu_utilization,
                    "capabilities": list(hb.capabilities),
                    "labels": dict(hb.labels),
                }
                for hb in workers.values()
            ],
        }
