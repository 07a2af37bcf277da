#!/usr/bin/env python3
"""Flagship benchmark: jobs dispatched/sec (whole node) + p50 dispatch->result
latency — the BASELINE.json metric on synthetic echo jobs.

Each timed step runs ONE full control-plane tick on every rank
(cordum_amd/ops/pipeline.py): submit -> K1 policy gate -> heartbeat
all-gather -> K2 least-loaded routing -> state transitions -> RCCL
all-to-all dispatch -> device echo workers -> result return -> SUCCEEDED.
Work is synthetic (random job descriptors + random 256 B payloads), nothing
is cached across steps (a ring of distinct pre-encoded batches), and every
job traverses the full state machine.

Usage (the driver contract):
  python bench.py [--gpus N] [--steps K] [--warmup W]
For N > 1 the driver launches via torch.distributed.run with one rank per
GPU; ranks read RANK/LOCAL_RANK/WORLD_SIZE from the env.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=16384, help="jobs per rank per step")
    ap.add_argument("--rules", type=int, default=1024, help="policy rules in the compiled bundle")
    ap.add_argument("--workers", type=int, default=1000, help="workers per rank")
    ap.add_argument("--payload-bytes", type=int, default=256)
    ap.add_argument("--allow-cpu", action="store_true",
                    help="CI validation: run the identical pipeline on CPU "
                         "(gloo + torch reference ops) — not a perf mode")
    args = ap.parse_args()

    import torch

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world_size == 1 and args.gpus > 1:
        print(
            "error: for --gpus N>1 launch via torch.distributed.run "
            "(one rank per GPU)",
            file=sys.stderr,
        )
        return 2

    use_gpu = torch.cuda.is_available()
    if not use_gpu and not args.allow_cpu:
        print("error: bench.py requires an MI355X (no GPU visible)", file=sys.stderr)
        return 1

    import torch.distributed as dist

    if world_size > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend="nccl" if use_gpu else "gloo",
                                rank=rank, world_size=world_size)
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
    else:
        device = torch.device("cpu")

    from cordum_amd.ops.pipeline import DevicePipeline

    pipe = DevicePipeline(
        device=device,
        batch_size=args.batch,
        n_local_workers=args.workers,
        n_rules=args.rules,
        payload_words=max(1, args.payload_bytes // 4),
        world_size=world_size,
        rank=rank,
        backend="ext" if use_gpu else "ref",
    )

    def barrier():
        if world_size > 1:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize(device)

    # warmup
    for _ in range(args.warmup):
        pipe.tick()
    barrier()

    if use_gpu:
        # sync-free timed window: stats fold into a device accumulator inside
        # the captured tick (read ONCE after the closing barrier), per-step
        # latency from hipEvents — the contract brackets the WHOLE window
        # with barrier+synchronize, not every step
        pipe.reset_stats()
        events = [torch.cuda.Event(enable_timing=True) for _ in range(args.steps + 1)]
        t0 = time.perf_counter()
        events[0].record()
        for s in range(args.steps):
            pipe.tick_async()
            events[s + 1].record()
        barrier()
        elapsed = time.perf_counter() - t0
        completed, denied = pipe.collect_stats()
        completed += denied  # denied jobs are also fully decided
        step_times = [events[s].elapsed_time(events[s + 1]) / 1000.0
                      for s in range(args.steps)]
    else:
        t0 = time.perf_counter()
        step_times = []
        completed = 0
        denied = 0
        for _ in range(args.steps):
            st = pipe.tick()
            step_times.append(st.wall_s)
            completed += st.completed + st.denied  # denied jobs also fully decided
            denied += st.denied
        barrier()
        elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks; SUM of completed jobs over ranks
    if world_size > 1:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        c = torch.tensor([completed], device=device, dtype=torch.float64)
        dist.all_reduce(c, op=dist.ReduceOp.SUM)
        completed = int(c.item())

    jobs_per_sec = completed / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    p50_ms = statistics.median(step_times) * 1000.0

    baseline_sustained = 38234.0  # BASELINE.md peak-stress sustained jobs/s
    if rank == 0:
        out = {
            "metric": "jobs dispatched/sec (whole node)",
            "value": round(jobs_per_sec, 1),
            "unit": "jobs/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(jobs_per_sec / baseline_sustained, 2),
            "dtype": "int64-bitset/int32 (control-plane integer path)",
            "data": "synthetic (random job descriptors + random 256B payloads, ring of distinct batches)",
            "config": {
                "model": "cordum control-plane dispatch pipeline (echo workers)",
                "global_batch": args.batch * world_size,
                "seq_len": args.payload_bytes,
                "parallelism": f"shard{world_size}",
                "rules": args.rules,
                "workers_per_rank": args.workers,
                "p50_dispatch_result_ms": round(p50_ms, 3),
                "denied_jobs": denied,
            },
        }
        print(json.dumps(out))
    if world_size > 1:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
