#!/usr/bin/env python3
"""Flagship benchmark: jobs dispatched/sec (whole node) + p50 dispatch->result
latency — the BASELINE.json metric on synthetic echo jobs.

Two timed windows, both reported in the ONE output line:

1. END-TO-END INGEST (the headline `value`): each step encodes a FRESH random
   job batch on the host (vectorized JobEncoder semantics, new content every
   step), H2D-copies descriptors + payload bytes into the device staging
   tensors, runs the full control-plane tick (K1 policy gate -> heartbeat
   fan-in -> K2 least-loaded routing -> state transitions -> dispatch ->
   device echo workers -> SUCCEEDED), and reads the per-job results
   (checksums + decisions) back to the host. Encode, H2D, tick and D2H are
   all inside the timed region.
2. IN-HBM TICK (`config.in_hbm_tick_jobs_per_s`): the steady-state dispatch
   tick over batches resident in HBM (a ring of pre-staged distinct batches,
   hipGraph-captured) — the rate of the dispatch engine itself once jobs are
   in device memory.

Counting: `value` counts DISPATCHED+COMPLETED jobs only; DENIED jobs are
reported separately in config.denied_jobs (they are decided, not dispatched).

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


def run_workflow_bench(args, device, world_size, rank, backend, use_gpu) -> int:
    """Config #3: waves of R runs per rank, each run = seed -> 1->fanout
    for_each -> approval gate (host-granted) -> final. One bench step = one
    full wave admitted and run to completion (run creation, every tick,
    collective exchanges and the approval host hop are all inside the timed
    region)."""
    import statistics

    import torch
    import torch.distributed as dist

    from cordum_amd.ops.wf_pipeline import DagSpec, WorkflowPipeline

    pipe = WorkflowPipeline(
        device=device,
        dags=[DagSpec.fanout_approval(args.fanout) for _ in range(args.runs)],
        n_local_workers=args.workers,
        payload_words=max(1, args.payload_bytes // 4),
        world_size=world_size,
        rank=rank,
        backend=backend,
        # arena sized for the whole wave: partial-emission backpressure is a
        # correctness feature, not a bench configuration
        child_cap=args.runs * (args.fanout + 2),
    )

    def barrier():
        if world_size > 1:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize(device)

    for _ in range(max(1, args.warmup)):
        pipe.run_wave()
    barrier()

    import gc

    gc.collect()
    gc.freeze()
    gc.disable()
    waves = []
    ok = fail = ticks = 0
    t0 = time.perf_counter()
    for _ in range(args.steps):
        st = pipe.run_wave()
        waves.append(st.wall_s)
        ok += st.runs_succeeded
        fail += st.runs_failed
        ticks += st.ticks
    barrier()
    elapsed = time.perf_counter() - t0
    gc.enable()

    if world_size > 1:
        t = torch.tensor([elapsed], device=device, dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        c = torch.tensor([float(ok)], device=device, dtype=torch.float64)
        dist.all_reduce(c)
        elapsed, ok_total = float(t.item()), int(c.item())
    else:
        ok_total = ok
    runs_per_s = ok_total / elapsed
    jobs_per_run = 2 + args.fanout  # seed + children + final
    if rank == 0:
        out = {
            "metric": "workflow runs/sec (config #3: 1->%d fan-out + approval)" % args.fanout,
            "value": round(runs_per_s, 1),
            "unit": "runs/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(runs_per_s * jobs_per_run / 8500.0, 2),
            "dtype": "int64-bitset/int32 (control-plane integer path)",
            "data": "synthetic (fresh run wave admitted per step; approval "
                    "granted by the host approver inside the timed region)",
            "config": {
                "model": "cordum device workflow engine (K3-WF tick)",
                "global_batch": args.runs * world_size,
                "seq_len": args.fanout,
                "parallelism": f"shard{world_size}",
                "runs_per_rank": args.runs,
                "fanout": args.fanout,
                "child_jobs_per_s": round(runs_per_s * jobs_per_run, 1),
                "avg_ticks_per_wave": round(ticks / max(1, args.steps), 1),
                "p50_wave_ms": round(statistics.median(waves) * 1000.0, 3),
                "runs_failed": fail,
                "vs_baseline_is": "child jobs/s vs reference workflow engine "
                                  "8,500 jobs/s (BASELINE.md)",
            },
        }
        print(json.dumps(out))
    if world_size > 1:
        dist.destroy_process_group()
    return 0


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
    ap.add_argument("--workflow", action="store_true",
                    help="config #3 mode: device workflow engine waves "
                         "(1->fanout fan-out + approval gate per run)")
    ap.add_argument("--runs", type=int, default=2048, help="workflow runs per rank per wave")
    ap.add_argument("--fanout", type=int, default=256, help="for_each children per run")
    args = ap.parse_args()

    import torch

    from cordum_amd.utils.threads import cap_torch_threads

    cap_torch_threads()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
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

    from cordum_amd.parallel import init_fabric

    rank, world_size, device = init_fabric()

    from cordum_amd.ops.pipeline import DevicePipeline

    backend = "ext" if use_gpu else "ref"

    if args.workflow:
        return run_workflow_bench(args, device, world_size, rank, backend, use_gpu)
    pipe = DevicePipeline(
        device=device,
        batch_size=args.batch,
        n_local_workers=args.workers,
        n_rules=args.rules,
        payload_words=max(1, args.payload_bytes // 4),
        world_size=world_size,
        rank=rank,
        backend=backend,
    )
    # e2e window pipeline: the captured graph reads the staging tensors the
    # per-step H2D copies write — MFMA K1 included, via the device-side job
    # pack kernel (falls back to the bitset K1 outside the MFMA envelope)
    pipe_e2e = DevicePipeline(
        device=device,
        batch_size=args.batch,
        n_local_workers=args.workers,
        n_rules=args.rules,
        payload_words=max(1, args.payload_bytes // 4),
        world_size=world_size,
        rank=rank,
        n_batches=2,
        backend=backend,
        mfma_pack_on_device=True,
    )

    def barrier():
        if world_size > 1:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize(device)

    # warmup both windows (graph capture, allocator steady state, RCCL init)
    for _ in range(args.warmup):
        pipe.tick()
    pipe_e2e.e2e_run(max(2, args.warmup))
    barrier()

    # GC discipline for the timed windows: the captured graphs + pipelines are
    # ~1M long-lived Python objects; a gen-2 collection mid-window is a ~90 ms
    # stall (measured, tools/e2e_breakdown.py). Freeze the steady state out of
    # the collector and disable automatic collection while timing — standard
    # latency-server practice, not a measurement trick (no work is skipped).
    import gc

    gc.collect()
    gc.freeze()
    gc.disable()

    # ---- window 1: end-to-end ingest (headline) --------------------------------
    # depth-2 pipelined on 1 GPU (encode t+1 on host while the device runs
    # tick t); the multi-rank and CPU paths run the sequential loop inside
    # e2e_run. Latencies are per-batch encode-start -> results-on-host.
    t0 = time.perf_counter()
    e2e_completed, e2e_denied, e2e_steps = pipe_e2e.e2e_run(args.steps)
    barrier()
    e2e_elapsed = time.perf_counter() - t0

    # ---- window 2: in-HBM steady-state tick -------------------------------------
    if use_gpu:
        # sync-free timed window: stats fold into a device accumulator inside
        # the captured tick (read ONCE after the closing barrier); per-step
        # latency from hipEvents
        pipe.reset_stats()
        events = [torch.cuda.Event(enable_timing=True) for _ in range(args.steps + 1)]
        t0 = time.perf_counter()
        events[0].record()
        for s in range(args.steps):
            pipe.tick_async()
            events[s + 1].record()
        barrier()
        tick_elapsed = time.perf_counter() - t0
        tick_completed, tick_denied = pipe.collect_stats()
        tick_steps = [events[s].elapsed_time(events[s + 1]) / 1000.0
                      for s in range(args.steps)]
    else:
        t0 = time.perf_counter()
        tick_steps = []
        tick_completed = 0
        tick_denied = 0
        for _ in range(args.steps):
            st = pipe.tick()
            tick_steps.append(st.wall_s)
            tick_completed += st.completed
            tick_denied += st.denied
        barrier()
        tick_elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks; SUM of completed jobs over ranks
    def reduce_window(elapsed, completed):
        if world_size == 1:
            return elapsed, completed
        t = torch.tensor([elapsed], device=device, dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        c = torch.tensor([completed], device=device, dtype=torch.float64)
        dist.all_reduce(c, op=dist.ReduceOp.SUM)
        return float(t.item()), int(c.item())

    gc.enable()
    e2e_elapsed, e2e_completed = reduce_window(e2e_elapsed, e2e_completed)
    tick_elapsed, tick_completed = reduce_window(tick_elapsed, tick_completed)

    e2e_jobs_per_sec = e2e_completed / e2e_elapsed
    tick_jobs_per_sec = tick_completed / tick_elapsed
    ms_per_step = e2e_elapsed / args.steps * 1000.0
    p50_ms = statistics.median(e2e_steps) * 1000.0

    baseline_sustained = 38234.0  # BASELINE.md peak-stress sustained jobs/s
    if rank == 0:
        out = {
            "metric": "jobs dispatched/sec (whole node)",
            "value": round(e2e_jobs_per_sec, 1),
            "unit": "jobs/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(e2e_jobs_per_sec / baseline_sustained, 2),
            "dtype": "int64-bitset/int32 (control-plane integer path)",
            "data": "synthetic (FRESH random job batch encoded+H2D per step; "
                    "payload bytes from a pre-generated pinned ring, stamped "
                    "per step and H2D-copied in the timed window)",
            "config": {
                "model": "cordum control-plane dispatch pipeline (echo workers)",
                "global_batch": args.batch * world_size,
                "seq_len": args.payload_bytes,
                "parallelism": f"shard{world_size}",
                "rules": args.rules,
                "workers_per_rank": args.workers,
                "value_is": "end-to-end ingest rate (host encode + H2D + full "
                            "tick + result D2H all timed)",
                "in_hbm_tick_jobs_per_s": round(tick_jobs_per_sec, 1),
                "in_hbm_tick_ms_per_step": round(tick_elapsed / args.steps * 1000.0, 3),
                "p50_dispatch_result_ms": round(p50_ms, 3),
                "denied_jobs": e2e_denied + tick_denied,
                "denied_excluded_from_value": True,
            },
        }
        print(json.dumps(out))
    if world_size > 1:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
