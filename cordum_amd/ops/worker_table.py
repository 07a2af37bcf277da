"""Device worker table: the registry snapshot packed into K2 scoring tensors.

The reference's least-loaded loop (strategy_least_loaded.go:92-119) walks a
host map per job; here the live worker set is packed once per flush into flat
device tensors (pool id, active, max_parallel, cpu, gpu, label mask) and the
K2 kernel scores W workers × J jobs in one launch (wave-per-job LDS argmin,
ops/hip/cordum_kernels.hip). The pack is cheap (one pass over ≤ thousands of
heartbeats) and reruns only when the registry snapshot version changes.

Encoding:
 - workers sorted by worker_id so the kernel's (score, idx) tie-break equals
   the host strategy's (score, worker_id) tie-break bit-for-bit;
 - pools interned to bit positions 0..62 of the job's int64 pool mask;
 - placement labels interned as "k=v" bits; a job label missing from every
   worker maps to the reserved IMPOSSIBLE bit (no worker carries it), which
   makes K2 return -1 = no_workers, same as the host filter;
 - >63 pools or >63 distinct label pairs -> exact=False and the caller falls
   back to the host strategy (correctness first; the fleet sizes the
   reference supports fit in one word).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from ..protocol.capv2 import Heartbeat

IMPOSSIBLE_BIT = 63  # set on jobs only: requires a label/pool no worker has


class DeviceWorkerTable:
    def __init__(self, device: torch.device, ext):
        self.device = torch.device(device)
        self.ext = ext
        self.worker_ids: List[str] = []
        self.pool_bit: Dict[str, int] = {}
        self.label_bit: Dict[str, int] = {}
        self.exact = True
        self.w_pool: Optional[torch.Tensor] = None
        self.w_keys: Optional[torch.Tensor] = None
        self.w_labels: Optional[torch.Tensor] = None
        self._packed_sig: Optional[tuple] = None

    # -- packing ---------------------------------------------------------------
    def pack(self, workers: Dict[str, Heartbeat]) -> bool:
        """Pack a registry snapshot; returns exact (False => host fallback).
        Skips repacking when the snapshot is unchanged (load vectors included:
        active_jobs feed the score)."""
        sig = tuple(
            (wid, hb.pool, hb.active_jobs, hb.max_parallel_jobs,
             round(hb.cpu_load, 3), round(hb.gpu_utilization, 3),
             tuple(sorted((hb.labels or {}).items())))
            for wid, hb in sorted(workers.items())
        )
        if sig == self._packed_sig:
            return self.exact
        self._packed_sig = sig

        ids = sorted(workers)
        self.worker_ids = ids
        self.pool_bit = {}
        self.label_bit = {}
        self.exact = True
        NW = len(ids)
        w_pool = torch.zeros(max(NW, 1), dtype=torch.int32)
        w_active = torch.zeros(max(NW, 1), dtype=torch.int32)
        w_maxp = torch.zeros(max(NW, 1), dtype=torch.int32)
        w_cpu = torch.zeros(max(NW, 1), dtype=torch.float32)
        w_gpu = torch.zeros(max(NW, 1), dtype=torch.float32)
        w_labels = torch.zeros(max(NW, 1), dtype=torch.int64)
        for i, wid in enumerate(ids):
            hb = workers[wid]
            pool = hb.pool or "default"
            b = self.pool_bit.get(pool)
            if b is None:
                if len(self.pool_bit) >= IMPOSSIBLE_BIT:
                    self.exact = False
                    b = IMPOSSIBLE_BIT - 1  # arbitrary; table unusable anyway
                else:
                    b = len(self.pool_bit)
                self.pool_bit[pool] = b
            w_pool[i] = b
            w_active[i] = int(hb.active_jobs)
            w_maxp[i] = int(hb.max_parallel_jobs)
            w_cpu[i] = float(hb.cpu_load)
            w_gpu[i] = float(hb.gpu_utilization)
            mask = 0
            for k, v in (hb.labels or {}).items():
                kv = f"{k}={v}"
                lb = self.label_bit.get(kv)
                if lb is None:
                    if len(self.label_bit) >= IMPOSSIBLE_BIT:
                        self.exact = False
                        continue
                    lb = len(self.label_bit)
                    self.label_bit[kv] = lb
                mask |= 1 << lb
            w_labels[i] = mask - (1 << 64) if mask >= (1 << 63) else mask

        dev = self.device
        self.w_pool = w_pool[:NW].to(dev) if NW else w_pool[:0].to(dev)
        self.w_labels = w_labels[:NW].to(dev) if NW else w_labels[:0].to(dev)
        if NW:
            self.w_keys = self.ext.worker_precompute(
                self.w_pool, w_active[:NW].to(dev), w_maxp[:NW].to(dev),
                w_cpu[:NW].to(dev), w_gpu[:NW].to(dev),
            )
        else:
            self.w_keys = torch.zeros(0, dtype=torch.int64, device=dev)
        return self.exact

    @property
    def n_workers(self) -> int:
        return len(self.worker_ids)

    # -- job-side encodes --------------------------------------------------------
    def pool_mask(self, pools: List[str]) -> int:
        m = 0
        known = False
        for p in pools:
            b = self.pool_bit.get(p)
            if b is not None:
                m |= 1 << b
                known = True
        if not known:
            m = 1 << IMPOSSIBLE_BIT  # no live worker in any eligible pool
        return m

    def label_mask(self, required: Dict[str, str]) -> int:
        m = 0
        for k, v in (required or {}).items():
            b = self.label_bit.get(f"{k}={v}")
            if b is None:
                return 1 << IMPOSSIBLE_BIT  # no worker carries this label
            m |= 1 << b
        return m

    # -- batched pick ------------------------------------------------------------
    def pick(self, pool_masks: List[int], label_masks: List[int]) -> torch.Tensor:
        """K2 over the packed table: int32 [J] worker idx, -1 no_workers,
        -2 pool_overloaded (the kernel's contract, tests/test_gpu_kernels.py)."""

        if self.n_workers == 0:
            return torch.full((len(pool_masks),), -1, dtype=torch.int32)

        def signed(vals):
            return torch.tensor(
                [v - (1 << 64) if v >= (1 << 63) else v for v in vals],
                dtype=torch.int64, device=self.device,
            )

        j_pool = signed(pool_masks)
        j_label = signed(label_masks)
        return self.ext.least_loaded_pick(self.w_pool, self.w_keys, self.w_labels, j_pool, j_label)

    def subject_for(self, idx: int) -> Tuple[str, str]:
        """(worker_id, direct subject) for a K2 pick index."""
        from ..protocol.subjects import worker_subject

        wid = self.worker_ids[idx]
        return wid, worker_subject(wid)
