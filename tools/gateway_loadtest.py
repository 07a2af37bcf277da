#!/usr/bin/env python3
"""Gateway load test: POST /api/v1/jobs throughput + latency percentiles
against a live `cordumctl serve` process (reference number: 5,234 req/s at
12.4 ms p99 on m5.2xlarge, BENCHMARKS.md §5 / BASELINE.md).

Plain keep-alive http.client connections on worker threads; every request
is a full job submission (validation, idempotency, secrets scan, context
write, meta persistence, dispatch). Reports a JSON line.

Usage: python tools/gateway_loadtest.py --seconds 10 --threads 16
"""
from __future__ import annotations

import argparse
import http.client
import json
import os
import signal
import socket
import statistics
import subprocess
import sys
import threading
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def worker(port: int, stop: threading.Event, lat: list, errs: list, tid: int):
    conn = http.client.HTTPConnection("127.0.0.1", port)
    body_tmpl = {"topic": "job.default", "prompt": "load test payload",
                 "labels": {"lt": "1"}}
    n = 0
    while not stop.is_set():
        body = dict(body_tmpl)
        body["idempotency_key"] = f"lt-{tid}-{n}"
        n += 1
        payload = json.dumps(body)
        t0 = time.perf_counter()
        try:
            conn.request("POST", "/api/v1/jobs", body=payload,
                         headers={"Content-Type": "application/json",
                                  "X-Principal-Id": "loadtest"})
            resp = conn.getresponse()
            resp.read()
            if resp.status != 200:
                errs.append(resp.status)
        except Exception as e:
            errs.append(str(e))
            conn = http.client.HTTPConnection("127.0.0.1", port)
            continue
        lat.append(time.perf_counter() - t0)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=10.0)
    ap.add_argument("--threads", type=int, default=16)
    ap.add_argument("--warmup", type=float, default=2.0)
    ap.add_argument("--dispatch", default="", help="override CORDUM_DISPATCH")
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    if args.dispatch:
        env["CORDUM_DISPATCH"] = args.dispatch
    proc = subprocess.Popen(
        [sys.executable, "-m", "cordum_amd.cli.cordumctl", "serve",
         "--port", str(port), "--workers", "4"],
        cwd=str(REPO), env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    )
    try:
        conn = None
        for _ in range(150):
            try:
                conn = http.client.HTTPConnection("127.0.0.1", port, timeout=2)
                conn.request("GET", "/api/v1/status",
                             headers={"X-Principal-Id": "loadtest"})
                conn.getresponse().read()
                break
            except Exception:
                if proc.poll() is not None:
                    print(proc.stdout.read().decode()[-2000:], file=sys.stderr)
                    return 1
                time.sleep(0.2)
        else:
            return 1

        stop = threading.Event()
        lat: list = []
        errs: list = []
        threads = [threading.Thread(target=worker, args=(port, stop, lat, errs, t),
                                    daemon=True) for t in range(args.threads)]
        for t in threads:
            t.start()
        time.sleep(args.warmup)
        lat.clear()
        n0 = len(lat)
        t0 = time.perf_counter()
        time.sleep(args.seconds)
        elapsed = time.perf_counter() - t0
        n1 = len(lat)
        stop.set()
        for t in threads:
            t.join(timeout=5)
        window = sorted(lat[n0:n1])
        rps = len(window) / elapsed
        out = {
            "benchmark": "POST /api/v1/jobs sustained (full submission path)",
            "req_per_s": round(rps, 1),
            "p50_ms": round(statistics.median(window) * 1000, 2) if window else None,
            "p99_ms": round(window[int(len(window) * 0.99)] * 1000, 2) if window else None,
            "threads": args.threads,
            "seconds": args.seconds,
            "errors": len(errs),
            "reference_req_per_s": 5234.0,
            "vs_reference": round(rps / 5234.0, 2),
            "dispatch": args.dispatch or "default",
        }
        print(json.dumps(out))
        if args.out:
            with open(args.out, "w") as f:
                json.dump(out, f, indent=1)
        return 0
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()


if __name__ == "__main__":
    sys.exit(main())
