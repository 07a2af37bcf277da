#!/usr/bin/env python3
"""BASELINE config #5: guardrails-style load at 1M concurrent jobs/runs with
retries/backoff + DLQ, sized for HBM3E (288 GB/GPU).

Allocates a 1M-slot HBM job table (states/attempts/deadlines/updated + a
payload arena), then drives reconciler-scale operations on it:

  - K4 deadline/staleness scan over the full 1M-slot table
  - TIMEOUT transitions for expired jobs (K5, legality-checked)
  - retry re-scheduling (TIMEOUT -> terminal is final in the job table; the
    workflow layer's retry policy creates attempt n+1 — modeled here by
    re-arming a fresh wave of slots with attempts+1 and new deadlines)
  - DLQ ring append for jobs whose attempts exceeded max_retries

Prints one JSON line with the measured per-op times and the projected
capacity at 288 GB. Run on a GPU box:
  python tools/scale_config5.py [--slots 1000000]
"""
from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import time


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--slots", type=int, default=1_000_000)
    ap.add_argument("--payload-bytes", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()

    import torch

    if not torch.cuda.is_available():
        print("error: needs a GPU", file=sys.stderr)
        return 1
    from cordum_amd.ops import get_ext
    from cordum_amd.protocol.states import transition_lut

    ext = get_ext(required=True)
    dev = torch.device("cuda:0")
    ext.set_transition_lut(torch.tensor(transition_lut(), dtype=torch.uint8).flatten())

    N = args.slots
    W = args.payload_bytes // 4
    g = torch.Generator().manual_seed(5)

    states = torch.randint(1, 6, (N,), dtype=torch.uint8, generator=g).to(dev)  # active states
    attempts = torch.randint(0, 3, (N,), dtype=torch.int32, generator=g).to(dev)
    now = 1_000_000_000
    deadlines = (now + torch.randint(-100_000, 1_000_000, (N,), dtype=torch.int64, generator=g)).to(dev)
    updated = (now - torch.randint(0, 600_000_000, (N,), dtype=torch.int64, generator=g)).to(dev)
    payload = torch.randint(-(1 << 31), (1 << 31) - 1, (N * W,), dtype=torch.int32, generator=g).to(dev)
    res_arena = torch.zeros_like(payload)
    res_sums = torch.zeros(N, dtype=torch.int32, device=dev)
    dlq_ring = torch.zeros(1 << 20, dtype=torch.int32, device=dev)
    dlq_head = torch.zeros(1, dtype=torch.int32, device=dev)

    torch.cuda.synchronize()

    def timed(fn, n=args.iters):
        fn()  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    # K4 full-table scan
    def scan():
        slots, count = ext.deadline_scan(states, deadlines, updated, now,
                                         now - 300_000_000, now - 500_000_000, N)
        return slots, count

    t_scan = timed(lambda: scan())
    slots, count = scan()
    torch.cuda.synchronize()
    n_expired = int(count.cpu()[0])

    # K5 transitions at scale: TIMEOUT the expired set
    expired_slots = slots[:min(n_expired, N)]
    to_timeout = torch.full((expired_slots.numel(),), 9, dtype=torch.uint8, device=dev)
    t_timeout = timed(lambda: ext.apply_transitions(states, attempts, deadlines,
                                                    expired_slots, to_timeout))

    # retry wave: re-arm 1/8 of the table as fresh SCHEDULED attempts
    retry_slots = torch.arange(0, N, 8, dtype=torch.int32, device=dev)
    def retry():
        states.index_fill_(0, retry_slots.long(), 1)  # back to PENDING (new attempt)
        ts = torch.full((retry_slots.numel(),), 3, dtype=torch.uint8, device=dev)
        ext.apply_transitions(states, attempts, deadlines, retry_slots, ts)
    t_retry = timed(retry)

    # DLQ ring append for exhausted jobs (attempts > 3)
    exhausted = (attempts > 3).nonzero().flatten().to(torch.int32)
    def dlq():
        n = exhausted.numel()
        if n:
            idx = (torch.arange(n, device=dev) + dlq_head) % dlq_ring.numel()
            dlq_ring[idx.long()] = exhausted
            dlq_head.add_(n)
    t_dlq = timed(dlq)

    # K3 readiness sweep over 1M runs (16-step workflows, dep chains)
    g2 = torch.Generator().manual_seed(7)
    step_state = torch.randint(0, 5, (N, 64), dtype=torch.uint8, generator=g2).to(dev)
    deps_mask = torch.zeros(N, 64, dtype=torch.int64, device=dev)
    deps_mask[:, 1:16] = (1 << torch.arange(15, device=dev)).view(1, 15)  # chain deps
    n_steps_t = torch.full((N,), 16, dtype=torch.uint8, device=dev)
    run_active = torch.ones(N, dtype=torch.uint8, device=dev)
    t_ready = timed(lambda: ext.run_readiness(step_state, deps_mask, n_steps_t, run_active, N))

    # payload touch at scale (1/16 of the arena per "tick")
    window = torch.arange(0, N, 16, dtype=torch.int32, device=dev)
    t_echo = timed(lambda: ext.echo_execute_indexed(payload, window, res_arena, res_sums, W))

    alloc = torch.cuda.memory_allocated(dev)
    bytes_per_job = alloc / N
    capacity_288g = int(288e9 / bytes_per_job)

    out = {
        "config": "baseline #5: 1M concurrent jobs w/ retries+DLQ, HBM sizing",
        "slots": N,
        "payload_bytes": args.payload_bytes,
        "hbm_allocated_gb": round(alloc / 1e9, 2),
        "bytes_per_job": round(bytes_per_job, 1),
        "projected_jobs_at_288GB": capacity_288g,
        "deadline_scan_1M_ms": round(t_scan * 1e3, 3),
        "scan_throughput_jobs_per_sec": round(N / t_scan),
        "timeout_transitions": n_expired,
        "timeout_transition_ms": round(t_timeout * 1e3, 3),
        "retry_wave_ms_125k": round(t_retry * 1e3, 3),
        "dlq_append_ms": round(t_dlq * 1e3, 3),
        "echo_62k_jobs_ms": round(t_echo * 1e3, 3),
        "readiness_sweep_1M_runs_ms": round(t_ready * 1e3, 3),
    }
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
