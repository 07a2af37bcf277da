"""Torch intra-op thread capping for cgroup-quota'd hosts.

GPU boxes in this pool expose 256 cores but cap the cgroup at ~16 CPUs
(/sys/fs/cgroup/cpu.max). torch's default pool = one spin-waiting OpenMP
thread per visible core, which periodically exhausts the CFS quota: the
process freezes ~88 ms of every 100 ms period (measured,
gpurun_out/e2e_break2 -> fixed in e2e_break3). Cap the pool to the quota.
"""
from __future__ import annotations

import os


def cgroup_cpu_quota() -> int:
    cpus = os.cpu_count() or 8
    try:
        with open("/sys/fs/cgroup/cpu.max") as f:
            quota, period = f.read().split()
            if quota != "max":
                return max(1, int(int(quota) / int(period)))
    except (OSError, ValueError):
        pass
    return cpus


def cap_torch_threads(max_threads: int = 16) -> int:
    import torch

    n = max(2, min(max_threads, cgroup_cpu_quota()))
    torch.set_num_threads(n)
    return n
