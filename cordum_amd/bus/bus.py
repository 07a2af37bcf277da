"""In-process message bus — the native replacement for NATS/JetStream.

The reference's control-plane traffic (core/infra/bus/nats.go:21-435) becomes
in-process dispatch in the single-node design: subjects, queue groups, NATS
wildcard matching (`*` one token, `>` tail), JetStream-style at-least-once
semantics for durable subjects (msg-id dedup with a 2-minute window
:404-434, NAK-with-delay redelivery :146-168), and best-effort delivery for
heartbeats/cancel/progress/workflow events.

Cross-GPU traffic does NOT go through this bus object: batched job
descriptors travel over RCCL all-to-all (ops/pipeline.py);
this bus carries the host-side control flow (gateway ↔ scheduler ↔ workflow
engine ↔ in-process workers) and is the seam the loopback tests use
(reference test seam: scheduler/integration_test.go:18-45).
"""
from __future__ import annotations

import heapq
import itertools
import threading
from functools import lru_cache
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from ..protocol import subjects as subj
from ..protocol.capv2 import BusPacket
from ..utils.clock import Clock, SYSTEM_CLOCK

BUS_MSG_ID_LABEL = "cordum.bus_msg_id"
DEDUP_WINDOW_S = 120.0
DEFAULT_MAX_DELIVER = 5


class RetryAfter(Exception):
    """Raised by a handler to NAK the message for redelivery after `delay_s`.

    Oracle: core/controlplane/scheduler/retry.go:1-47 + bus NAK handling."""

    def __init__(self, delay_s: float, cause: str = ""):
        super().__init__(f"retry after {delay_s}s: {cause}")
        self.delay_s = delay_s
        self.cause = cause


@lru_cache(maxsize=8192)
def _tokens(s: str):
    return s.split(".")


def subject_matches(pattern: str, subject: str) -> bool:
    """NATS subject matching: tokens split on '.', '*' = one token, '>' = tail."""
    if pattern == subject:
        return True
    pt = _tokens(pattern)
    st = _tokens(subject)
    for i, p in enumerate(pt):
        if p == ">":
            return len(st) > i  # NATS '>': one or MORE remaining tokens
        if i >= len(st):
            return False
        if p != "*" and p != st[i]:
            return False
    return len(pt) == len(st)


def compute_msg_id(subject: str, pkt: BusPacket) -> Optional[str]:
    """bus/nats.go:404-434: explicit label wins; job requests 'jobreq:<id>';
    other job-bearing packets 'subject:<jobid>'; heartbeats none."""
    override = pkt.labels.get(BUS_MSG_ID_LABEL, "")
    if not override and pkt.job_request is not None:
        # the gateway sets the override on the job request's labels
        # (approval republish gateway.go:3785-3797, remediation :1594-1755)
        override = pkt.job_request.labels.get(BUS_MSG_ID_LABEL, "")
    if override:
        return override
    if pkt.heartbeat is not None:
        return None
    if pkt.job_request is not None and pkt.job_request.job_id:
        return f"jobreq:{pkt.job_request.job_id}"
    jid = pkt.job_id()
    if jid:
        return f"{subject}:{jid}"
    return None


@dataclass
class Subscription:
    pattern: str
    handler: Callable[[str, BusPacket], None]
    queue_group: Optional[str] = None
    id: int = 0
    active: bool = True
    deferred: bool = False  # deliver via pump() instead of inline (worker-style consumer)

    def unsubscribe(self):
        self.active = False


@dataclass(order=True)
class _Pending:
    due: float
    seq: int
    subject: str = field(compare=False)
    pkt: BusPacket = field(compare=False)
    deliveries: int = field(compare=False, default=0)
    msg_id: Optional[str] = field(compare=False, default=None)
    target: Optional[Subscription] = field(compare=False, default=None)


class Bus:
    """Abstract bus interface (scheduler/types.go:12-133 Bus)."""

    def publish(self, subject: str, pkt: BusPacket) -> None:
        raise NotImplementedError

    def subscribe(self, pattern: str, handler, queue_group: Optional[str] = None) -> Subscription:
        raise NotImplementedError

    def direct_subject(self, worker_id: str) -> str:
        return subj.worker_subject(worker_id)


class LoopbackBus(Bus):
    """Synchronous in-process bus with JetStream-equivalent semantics.

    Delivery is synchronous inside `publish` (deterministic for tests and for
    the single-process tick loop). A handler raising RetryAfter re-enqueues
    the message; `pump()` delivers messages whose delay has elapsed.
    """

    def __init__(self, clock: Clock = SYSTEM_CLOCK, max_deliver: int = DEFAULT_MAX_DELIVER):
        self._clock = clock
        self._mu = threading.RLock()
        self._subs: List[Subscription] = []
        self._rr: Dict[Tuple[str, str], int] = {}  # (group, pattern) round-robin index
        self._sub_seq = itertools.count(1)
        self._seq = itertools.count(1)
        self._dedup: Dict[str, float] = {}
        self._delayed: List[_Pending] = []
        self._max_deliver = max_deliver
        self.published: List[Tuple[str, BusPacket]] = []  # inspection hook (tests)
        self.record_published = False

    # -- subscribe -----------------------------------------------------------
    def subscribe(
        self, pattern: str, handler, queue_group: Optional[str] = None, deferred: bool = False
    ) -> Subscription:
        sub = Subscription(pattern, handler, queue_group, id=next(self._sub_seq), deferred=deferred)
        with self._mu:
            self._subs.append(sub)
        return sub

    # -- publish -------------------------------------------------------------
    def publish(self, subject: str, pkt: BusPacket) -> None:
        if subj.is_durable_subject(subject):
            msg_id = compute_msg_id(subject, pkt)
            if msg_id is not None:
                # dedup is per-stream, like JetStream's CORDUM_SYS (sys.>) vs
                # CORDUM_JOBS (job.>, worker.*.jobs) split (bus/nats.go:339-361):
                # the same jobreq:<id> msg-id may legally appear on both.
                stream = "SYS" if subject.startswith("sys.") else "JOBS"
                msg_id = f"{stream}:{msg_id}"
                if self._seen(msg_id):
                    return
        else:
            msg_id = None
        if self.record_published:
            self.published.append((subject, pkt))
        self._deliver(subject, pkt, deliveries=0, msg_id=msg_id)

    def publish_after(self, delay_s: float, subject: str, pkt: BusPacket) -> None:
        with self._mu:
            heapq.heappush(
                self._delayed,
                _Pending(self._clock.now() + delay_s, next(self._seq), subject, pkt),
            )

    def _seen(self, msg_id: str) -> bool:
        now = self._clock.now()
        with self._mu:
            t = self._dedup.get(msg_id)
            if t is not None and t > now - DEDUP_WINDOW_S:
                return True
            # amortized-O(1) expiry: insertion order IS time order, so pop
            # expired entries from the front (the old full-dict rebuild per
            # publish was O(n^2) under sustained load — the gateway load
            # test showed it as 75% of the whole submit path)
            cutoff = now - DEDUP_WINDOW_S
            while self._dedup:
                k = next(iter(self._dedup))
                if self._dedup[k] > cutoff and len(self._dedup) <= (1 << 18):
                    break
                del self._dedup[k]
            if t is not None:
                # expired: re-insert at the tail (may already be gone if the
                # front-expiry loop just removed it)
                self._dedup.pop(msg_id, None)
            self._dedup[msg_id] = now
            return False

    def _targets(self, subject: str) -> List[Subscription]:
        with self._mu:
            self._subs = [s for s in self._subs if s.active]
            matched = [s for s in self._subs if subject_matches(s.pattern, subject)]
            out: List[Subscription] = []
            groups: Dict[Tuple[str, str], List[Subscription]] = {}
            for s in matched:
                if s.queue_group is None:
                    out.append(s)
                else:
                    groups.setdefault((s.queue_group, s.pattern), []).append(s)
            for key, members in groups.items():
                idx = self._rr.get(key, 0)
                out.append(members[idx % len(members)])
                self._rr[key] = idx + 1
            return out

    def _deliver(
        self,
        subject: str,
        pkt: BusPacket,
        deliveries: int,
        msg_id: Optional[str],
        only: Optional[Subscription] = None,
    ) -> None:
        targets = [only] if only is not None else self._targets(subject)
        for s in targets:
            if not s.active:
                continue
            if s.deferred and only is None:
                with self._mu:
                    heapq.heappush(
                        self._delayed,
                        _Pending(self._clock.now(), next(self._seq), subject, pkt, deliveries, msg_id, s),
                    )
                continue
            try:
                s.handler(subject, pkt)
            except RetryAfter as ra:
                if subj.is_durable_subject(subject) and deliveries + 1 < self._max_deliver:
                    with self._mu:
                        heapq.heappush(
                            self._delayed,
                            _Pending(
                                self._clock.now() + ra.delay_s,
                                next(self._seq),
                                subject,
                                pkt,
                                deliveries + 1,
                                msg_id,
                                s,  # redeliver to the NAKing consumer only
                            ),
                        )
            # other exceptions propagate: handlers are expected to be total

    # -- pump delayed/redelivery queue ----------------------------------------
    def pump(self, now: Optional[float] = None) -> int:
        """Deliver all due delayed messages; returns count delivered."""
        t = now if now is not None else self._clock.now()
        n = 0
        while True:
            with self._mu:
                if not self._delayed or self._delayed[0].due > t:
                    return n
                item = heapq.heappop(self._delayed)
            self._deliver(item.subject, item.pkt, item.deliveries, item.msg_id, only=item.target)
            n += 1

    def pending_count(self) -> int:
        with self._mu:
            return len(self._delayed)

    def next_due(self) -> Optional[float]:
        with self._mu:
            return self._delayed[0].due if self._delayed else None
