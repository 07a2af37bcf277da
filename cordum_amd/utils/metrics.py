"""Prometheus metrics with the reference's metric names.

Oracle: core/infra/metrics/metrics.go — scheduler counters
cordum_scheduler_jobs_received_total{topic} / jobs_dispatched_total{topic} /
jobs_completed_total{topic,status} / safety_denied_total{topic} (:39-100),
gateway http_requests_total + request latency histogram (:105-135), workflow
started/completed/duration (:139-180), plus a Noop implementation (:31-37).
Exposition uses prometheus_client when importable, else a built-in text
renderer with the same names.
"""
from __future__ import annotations

import threading
from typing import Dict, Tuple

try:
    from prometheus_client import CollectorRegistry, Counter, Histogram, generate_latest

    HAVE_PROM = True
except Exception:  # pragma: no cover
    HAVE_PROM = False


class Metrics:
    def __init__(self):
        if HAVE_PROM:
            self.registry = CollectorRegistry()
            self.jobs_received = Counter(
                "cordum_scheduler_jobs_received_total", "Jobs received", ["topic"], registry=self.registry)
            self.jobs_dispatched = Counter(
                "cordum_scheduler_jobs_dispatched_total", "Jobs dispatched", ["topic"], registry=self.registry)
            self.jobs_completed = Counter(
                "cordum_scheduler_jobs_completed_total", "Jobs completed", ["topic", "status"], registry=self.registry)
            self.safety_denied = Counter(
                "cordum_scheduler_safety_denied_total", "Safety denied", ["topic"], registry=self.registry)
            self.http_requests = Counter(
                "cordum_gateway_http_requests_total", "HTTP requests", ["method", "path", "code"], registry=self.registry)
            self.http_latency = Histogram(
                "cordum_gateway_http_request_duration_seconds", "HTTP latency", ["method", "path"], registry=self.registry)
            self.workflows_started = Counter(
                "cordum_workflow_runs_started_total", "Workflow runs started", ["workflow"], registry=self.registry)
            self.workflows_completed = Counter(
                "cordum_workflow_runs_completed_total", "Workflow runs completed", ["workflow", "status"], registry=self.registry)
        else:
            self._mu = threading.Lock()
            self._counters: Dict[Tuple[str, Tuple[str, ...]], float] = {}

    # -- scheduler.Metrics interface -----------------------------------------
    def inc_received(self, topic: str):
        self._inc("cordum_scheduler_jobs_received_total", topic)

    def inc_dispatched(self, topic: str):
        self._inc("cordum_scheduler_jobs_dispatched_total", topic)

    def inc_completed(self, topic: str, status: str):
        self._inc("cordum_scheduler_jobs_completed_total", topic, status)

    def inc_safety_denied(self, topic: str):
        self._inc("cordum_scheduler_safety_denied_total", topic)

    def _inc(self, name: str, *labels: str):
        if HAVE_PROM:
            metric = {
                "cordum_scheduler_jobs_received_total": self.jobs_received,
                "cordum_scheduler_jobs_dispatched_total": self.jobs_dispatched,
                "cordum_scheduler_jobs_completed_total": self.jobs_completed,
                "cordum_scheduler_safety_denied_total": self.safety_denied,
            }[name]
            metric.labels(*labels).inc()
        else:
            with self._mu:
                key = (name, labels)
                self._counters[key] = self._counters.get(key, 0) + 1

    def exposition(self) -> bytes:
        if HAVE_PROM:
            return generate_latest(self.registry)
        lines = []
        with self._mu:
            for (name, labels), v in sorted(self._counters.items()):
                label_str = ",".join(f'l{i}="{x}"' for i, x in enumerate(labels))
                lines.append(f"{name}{{{label_str}}} {v}")
        return ("\n".join(lines) + "\n").encode()


class NoopMetrics:
    def inc_received(self, topic):
        pass

    def inc_dispatched(self, topic):
        pass

    def inc_completed(self, topic, status):
        pass

    def inc_safety_denied(self, topic):
        pass

    def exposition(self) -> bytes:
        return b""
