"""Scheduling strategies: least-loaded (capability/placement aware) + naive.

Semantics oracle: scheduler/strategy_least_loaded.go:40-275 —
 - topic -> pools routing (multi-pool), preferred_pool hint must be mapped
 - `requires` capability filter against pool profiles (all-of, case-insensitive)
 - placement-label filter (traceability labels excluded :196-224)
 - preferred_worker_id shortcut when healthy
 - overload skip: active/max_parallel >= 0.9, or cpu>=90, or gpu>=90
 - score = active_jobs + cpu/100 + gpu/100, lowest wins
 - direct subject `worker.<id>.jobs`, topic fallback
 - atomic routing hot-swap (UpdateRouting :28-38)

Tie-break: the Go implementation iterates a map (unspecified order); this
implementation breaks ties deterministically by worker_id, and the HIP scoring
kernel (ops/hip/score_kernels.hip, K2) implements the identical
(score, worker_idx) argmin so host and device picks agree bit-for-bit.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..protocol.capv2 import Heartbeat, JobRequest
from ..protocol.subjects import worker_subject
from . import errors as errs

OVERLOAD_UTILIZATION_THRESHOLD = 0.9

_NON_PLACEMENT_LABELS = {
    "preferred_worker_id",
    "preferred_pool",
    "secrets_present",
    "workflow_id",
    "run_id",
    "step_id",
    "node_id",
    "worker_id",
    # gateway-stamped audit/traceability labels (remediation
    # gateway.go:1594-1755, DLQ retry :3452-3553): in the reference these
    # leak into placement and make every remediated/retried job unroutable
    # (no worker carries them) — the same bug family as the approval_*
    # labels below. Audit metadata must not constrain placement.
    "remediation_of",
    "remediation_id",
    "retry",
    "dlq_entry",
    "retry_of_job",
}
# NOTE: the Go filter (strategy_least_loaded.go:196-224) drops only
# "approval_granted"; the gateway also stamps approval_reason/approval_note on
# approval (gateway.go:3788-3796), which there become placement constraints
# that no worker carries. We drop the whole approval_* family — these are
# audit metadata, not placement labels.


@dataclass
class PoolProfile:
    requires: List[str] = field(default_factory=list)


@dataclass
class PoolRouting:
    topics: Dict[str, List[str]] = field(default_factory=dict)
    pools: Dict[str, PoolProfile] = field(default_factory=dict)

    def clone(self) -> "PoolRouting":
        return PoolRouting(
            topics={t: list(p) for t, p in self.topics.items()},
            pools={n: PoolProfile(list(p.requires)) for n, p in self.pools.items()},
        )


def filter_placement_labels(labels: Dict[str, str]) -> Dict[str, str]:
    out = {}
    for k, v in (labels or {}).items():
        if k in _NON_PLACEMENT_LABELS or k.startswith("cordum.") or k.startswith("approval_"):
            continue
        out[k] = v
    return out


def load_score(hb: Heartbeat) -> float:
    return float(hb.active_jobs) + hb.cpu_load / 100.0 + hb.gpu_utilization / 100.0


def is_overloaded(hb: Heartbeat) -> bool:
    if hb.max_parallel_jobs > 0:
        if hb.active_jobs / hb.max_parallel_jobs >= OVERLOAD_UTILIZATION_THRESHOLD:
            return True
    if hb.cpu_load >= 90:
        return True
    if hb.gpu_utilization >= 90:
        return True
    return False


def matches_labels(hb: Heartbeat, required: Dict[str, str]) -> bool:
    if not required:
        return True
    if not hb.labels:
        return False
    return all(hb.labels.get(k) == v for k, v in required.items())


def pool_satisfies(pool_requires: List[str], job_requires: List[str]) -> bool:
    if not job_requires:
        return True
    if not pool_requires:
        return False
    have = {r.strip().lower() for r in pool_requires if r.strip()}
    return all(r.strip().lower() in have for r in job_requires if r.strip())


class Strategy:
    def pick_subject(self, req: JobRequest, workers: Dict[str, Heartbeat]) -> str:
        raise NotImplementedError


class NaiveStrategy(Strategy):
    """Pass-through topic -> subject (strategy_naive.go:10-21)."""

    def pick_subject(self, req: JobRequest, workers: Dict[str, Heartbeat]) -> str:
        if not req.topic:
            raise errs.NoPoolMapping("missing topic")
        return req.topic


class LeastLoadedStrategy(Strategy):
    def __init__(self, routing: Optional[PoolRouting] = None):
        self._mu = threading.Lock()
        self._routing = (routing or PoolRouting()).clone()

    def update_routing(self, routing: PoolRouting) -> None:
        with self._mu:
            self._routing = routing.clone()

    def current_routing(self) -> PoolRouting:
        with self._mu:
            return self._routing.clone()

    def resolve(self, req: JobRequest) -> Tuple[List[str], Dict[str, str], str]:
        """The host-cheap half of pick_subject: topic->pool resolution,
        preferred_pool hint, `requires` filtering, placement-label extraction.
        Returns (eligible_pools, required_labels, preferred_worker_id); raises
        NoPoolMapping exactly where pick_subject would. The scoring half runs
        either in the host loop below or in the K2 device kernel
        (ops/worker_table.py) over the same resolution."""
        if req is None or not req.topic:
            raise errs.NoPoolMapping("missing topic")
        routing = self.current_routing()
        labels = req.labels or {}
        required_labels = filter_placement_labels(labels)

        topic_pools = list(routing.topics.get(req.topic, []))
        pool_hint = labels.get("preferred_pool", "")
        if pool_hint:
            if pool_hint not in topic_pools:
                raise errs.NoPoolMapping(
                    f"preferred pool {pool_hint!r} not mapped for topic {req.topic!r}"
                )
            topic_pools = [pool_hint]
        if not topic_pools:
            raise errs.NoPoolMapping(f"topic {req.topic!r}")

        job_requires = list(req.meta.requires) if req.meta else []
        eligible = [
            p
            for p in topic_pools
            if not job_requires or pool_satisfies(routing.pools.get(p, PoolProfile()).requires, job_requires)
        ]
        if not eligible:
            raise errs.NoPoolMapping("no pool satisfies requires")
        return eligible, required_labels, labels.get("preferred_worker_id", "")

    def pick_subject(self, req: JobRequest, workers: Dict[str, Heartbeat]) -> str:
        eligible, required_labels, preferred = self.resolve(req)
        pool_set = set(eligible)

        if preferred:
            hb = workers.get(preferred)
            if hb is not None and hb.pool in pool_set and matches_labels(hb, required_labels) and not is_overloaded(hb):
                return worker_subject(preferred)

        selected: Optional[Heartbeat] = None
        best: Tuple[float, str] = (0.0, "")
        overloaded = 0
        total = 0
        for wid in sorted(workers):
            hb = workers[wid]
            if hb is None or hb.pool not in pool_set:
                continue
            if not matches_labels(hb, required_labels):
                continue
            total += 1
            if is_overloaded(hb):
                overloaded += 1
                continue
            key = (load_score(hb), wid)
            if selected is None or key < best:
                selected = hb
                best = key

        if selected is None:
            pools_str = ",".join(eligible)
            if total > 0 and overloaded == total:
                raise errs.PoolOverloaded(f"pool {pools_str!r}")
            raise errs.NoWorkers(f"pool {pools_str!r}")
        return worker_subject(selected.worker_id)


def routing_from_pools_yaml(doc: dict) -> PoolRouting:
    """Parse the reference's pools.yaml format (infra/config/pools.go:13-130):
    topics: {topic: pool | [pools]}, pools: {name: {requires: [...]}}."""
    topics: Dict[str, List[str]] = {}
    for topic, val in (doc.get("topics", {}) or {}).items():
        if isinstance(val, str):
            topics[topic] = [val]
        elif isinstance(val, list):
            topics[topic] = [str(v) for v in val]
        elif isinstance(val, dict) and "pools" in val:
            topics[topic] = [str(v) for v in val["pools"]]
    pools: Dict[str, PoolProfile] = {}
    for name, val in (doc.get("pools", {}) or {}).items():
        reqs = []
        if isinstance(val, dict):
            reqs = [str(r) for r in (val.get("requires", []) or [])]
        pools[name] = PoolProfile(requires=reqs)
    return PoolRouting(topics=topics, pools=pools)
