#!/usr/bin/env python3
"""Sustained-load soak (reference analog: BENCHMARKS.md §2 '8h sustained'):
run the flagship tick continuously, report sustained jobs/s, p50/p99 step
latency, and host RSS growth."""
import argparse
import json
import resource
from array import array
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=60.0)
    ap.add_argument("--batch", type=int, default=16384)
    args = ap.parse_args()
    import torch

    if not torch.cuda.is_available():
        print("needs GPU", file=sys.stderr)
        return 1
    from cordum_amd.ops.pipeline import DevicePipeline

    pipe = DevicePipeline(device=torch.device("cuda:0"), batch_size=args.batch)
    for _ in range(10):
        pipe.tick()
    rss0 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
    # production mode: sync-free ticks in windows of 256, stats from the
    # device accumulator, per-step latency via hipEvents
    pipe.reset_stats()
    times = array("d")  # compact: 2.6M python floats read as RSS "growth"
    completed = 0
    WINDOW = 256
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.seconds:
        events = [torch.cuda.Event(enable_timing=True) for _ in range(WINDOW + 1)]
        events[0].record()
        for s_ in range(WINDOW):
            pipe.tick_async()
            events[s_ + 1].record()
        torch.cuda.synchronize()
        times.extend(events[k].elapsed_time(events[k + 1]) / 1e3 for k in range(WINDOW))
    elapsed = time.perf_counter() - t0
    c, d = pipe.collect_stats()
    completed = c + d
    rss1 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
    times = sorted(times)
    print(json.dumps({
        "soak_seconds": round(elapsed, 1),
        "steps": len(times),
        "jobs_completed": completed,
        "sustained_jobs_per_sec": round(completed / elapsed),
        "p50_ms": round(times[len(times) // 2] * 1e3, 3),
        "p99_ms": round(times[int(len(times) * 0.99)] * 1e3, 3),
        "max_ms": round(times[-1] * 1e3, 3),
        "rss_growth_mb": round((rss1 - rss0) / 1024, 1),
    }))
    return 0


if __name__ == "__main__":
    sys.exit(main())
