"""Policy compiler: SafetyPolicy rules -> flat int tensors for the K1 kernel.

Lowers the first-match evaluator (safety/policy.py, oracle
safety_policy.go:187-294) into bitset tensors:

 - 7 "any-of" dimensions (tenant, topic-pattern, capability, pack, actor_id,
   actor_type, risk_tags): rule mask empty = unconstrained, else the job mask
   must intersect it;
 - 2 "all-of" dimensions (requires, labels): every rule bit must be present
   in the job mask;
 - secrets_present tri-state byte;
 - MCP allow/deny masks for 4 fields (server/tool/resource/action) with the
   "deny wins, non-empty allow is exclusive, unused request passes" logic.

Glob topic patterns stay EXACT: the job encoder computes each distinct
topic's pattern-membership mask on the host with the real path.Match
implementation (utils/globmatch.py) and caches it, so the device kernel only
ANDs bitmasks. Policies whose vocabularies overflow the configured word
count compile with exact=False and the pipeline uses the host evaluator.

Layout (R rules, W int64 words/dimension):
  any_masks   int64 [R, 7, W]
  all_masks   int64 [R, 2, W]
  secrets     int8  [R]        (-1 unconstrained / 0 / 1)
  mcp_masks   int64 [R, 4, 2, W]  (field, 0=allow 1=deny)
  mcp_any     uint8 [R]        (rule has any MCP constraint)
  decisions   int8  [R]        (DecisionType enum value)
Job batch (J jobs):
  any_bits    int64 [J, 7, W]
  all_bits    int64 [J, 2, W]
  secrets     uint8 [J]
  mcp_bits    int64 [J, 4, W]
  mcp_used    uint8 [J]
First match = min rule index whose every dimension matches; -1 = default
allow (policy.evaluate's fallthrough).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import torch

from ..safety import policy as pol
from ..utils.globmatch import topic_matches
from .vocab import Interner, bit_mask, single_bit

N_ANY = 7  # tenant, topic, capability, pack, actor_id, actor_type, risk
N_ALL = 2  # requires, labels
DIM_TENANT, DIM_TOPIC, DIM_CAP, DIM_PACK, DIM_ACTOR, DIM_ATYPE, DIM_RISK = range(7)
ALL_REQUIRES, ALL_LABELS = 0, 1
MCP_FIELDS = ("server", "tool", "resource", "action")

DECISION_CODE = {
    pol.DECISION_ALLOW: 1,
    pol.DECISION_DENY: 2,
    pol.DECISION_REQUIRE_APPROVAL: 3,
    pol.DECISION_THROTTLE: 4,
    pol.DECISION_ALLOW_WITH_CONSTRAINTS: 5,
}


@dataclass
class CompiledPolicy:
    words: int
    rules: List[pol.PolicyRule]
    any_masks: torch.Tensor
    all_masks: torch.Tensor
    secrets: torch.Tensor
    mcp_masks: torch.Tensor
    mcp_any: torch.Tensor
    decisions: torch.Tensor
    # vocabularies (shared with the job encoder)
    tenants: Interner = None
    patterns: List[str] = field(default_factory=list)
    caps: Interner = None
    packs: Interner = None
    actors: Interner = None
    atypes: Interner = None
    risks: Interner = None
    requires: Interner = None
    labels: Interner = None
    mcp: Interner = None
    exact: bool = True
    snapshot: str = ""

    @property
    def n_rules(self) -> int:
        return int(self.any_masks.shape[0])

    def to(self, device) -> "CompiledPolicy":
        clone = CompiledPolicy(
            words=self.words,
            rules=self.rules,
            any_masks=self.any_masks.to(device),
            all_masks=self.all_masks.to(device),
            secrets=self.secrets.to(device),
            mcp_masks=self.mcp_masks.to(device),
            mcp_any=self.mcp_any.to(device),
            decisions=self.decisions.to(device),
            tenants=self.tenants,
            patterns=self.patterns,
            caps=self.caps,
            packs=self.packs,
            actors=self.actors,
            atypes=self.atypes,
            risks=self.risks,
            requires=self.requires,
            labels=self.labels,
            mcp=self.mcp,
            exact=self.exact,
            snapshot=self.snapshot,
        )
        return clone


def compile_policy(policy: Optional[pol.SafetyPolicy], words: int = 1, snapshot: str = "") -> CompiledPolicy:
    rules = policy.effective_rules() if policy is not None else []
    R = len(rules)
    cap = 64 * words
    tenants = Interner(cap)
    caps = Interner(cap)
    packs = Interner(cap)
    actors = Interner(cap)
    atypes = Interner(cap)
    risks = Interner(cap)
    requires = Interner(cap)
    labels = Interner(cap, casefold=False)
    mcp = Interner(cap)
    patterns: List[str] = []
    pattern_ids: Dict[str, int] = {}

    any_masks = torch.zeros((max(R, 1), N_ANY, words), dtype=torch.int64)
    all_masks = torch.zeros((max(R, 1), N_ALL, words), dtype=torch.int64)
    secrets = torch.full((max(R, 1),), -1, dtype=torch.int8)
    mcp_masks = torch.zeros((max(R, 1), 4, 2, words), dtype=torch.int64)
    mcp_any = torch.zeros((max(R, 1),), dtype=torch.uint8)
    decisions = torch.zeros((max(R, 1),), dtype=torch.int8)

    exact = True

    def put_any(r: int, dim: int, ids: Sequence[int]):
        m = bit_mask(ids, words)
        for w in range(words):
            any_masks[r, dim, w] = m[w]

    def put_all(r: int, dim: int, ids: Sequence[int]):
        m = bit_mask(ids, words)
        for w in range(words):
            all_masks[r, dim, w] = m[w]

    for r, rule in enumerate(rules):
        m = rule.match
        put_any(r, DIM_TENANT, [tenants.intern(t) for t in m.tenants])
        tids = []
        for p in m.topics:
            p2 = p.strip()
            if not p2:
                continue
            pid = pattern_ids.get(p2)
            if pid is None:
                if len(patterns) >= cap:
                    exact = False
                    continue
                pid = len(patterns)
                pattern_ids[p2] = pid
                patterns.append(p2)
            tids.append(pid)
        put_any(r, DIM_TOPIC, tids)
        # a rule constraining topics that all overflowed must not silently
        # become unconstrained
        if m.topics and not tids:
            exact = False
        put_any(r, DIM_CAP, [caps.intern(c) for c in m.capabilities])
        put_any(r, DIM_PACK, [packs.intern(p) for p in m.pack_ids])
        put_any(r, DIM_ACTOR, [actors.intern(a) for a in m.actor_ids])
        put_any(r, DIM_ATYPE, [atypes.intern(a) for a in m.actor_types])
        put_any(r, DIM_RISK, [risks.intern(t) for t in m.risk_tags])
        if m.tenants and all(tenants.lookup(t) < 0 for t in m.tenants):
            exact = False
        put_all(r, ALL_REQUIRES, [requires.intern(t) for t in m.requires])
        put_all(r, ALL_LABELS, [labels.intern(f"{k}={v}") for k, v in sorted(m.labels.items())])
        if m.secrets_present is not None:
            secrets[r] = 1 if m.secrets_present else 0
        rule_mcp = m.mcp
        has_mcp = not rule_mcp.is_empty()
        mcp_any[r] = 1 if has_mcp else 0
        if has_mcp:
            for f, fname in enumerate(MCP_FIELDS):
                allow = getattr(rule_mcp, f"allow_{fname}s")
                deny = getattr(rule_mcp, f"deny_{fname}s")
                am = bit_mask([mcp.intern(v) for v in allow], words)
                dm = bit_mask([mcp.intern(v) for v in deny], words)
                for w in range(words):
                    mcp_masks[r, f, 0, w] = am[w]
                    mcp_masks[r, f, 1, w] = dm[w]
                if (allow and all(mcp.lookup(v) < 0 for v in allow)) or (
                    deny and all(mcp.lookup(v) < 0 for v in deny)
                ):
                    exact = False
        decisions[r] = DECISION_CODE.get(pol.normalize_decision(rule.decision), 1)
        # overflow in any-of dims with constraints that vanished -> inexact
        for interner, values in (
            (caps, m.capabilities), (packs, m.pack_ids), (actors, m.actor_ids),
            (atypes, m.actor_types), (risks, m.risk_tags),
        ):
            if values and all(interner.lookup(v) < 0 for v in values):
                exact = False
        if m.requires and interner_overflowed(requires, m.requires):
            exact = False
        if m.labels and any(labels.lookup(f"{k}={v}") < 0 for k, v in m.labels.items()):
            exact = False

    for it in (tenants, caps, packs, actors, atypes, risks, requires, labels, mcp):
        if it.overflow:
            exact = False

    return CompiledPolicy(
        words=words,
        rules=rules,
        any_masks=any_masks[:R] if R else any_masks[:0],
        all_masks=all_masks[:R] if R else all_masks[:0],
        secrets=secrets[:R] if R else secrets[:0],
        mcp_masks=mcp_masks[:R] if R else mcp_masks[:0],
        mcp_any=mcp_any[:R] if R else mcp_any[:0],
        decisions=decisions[:R] if R else decisions[:0],
        tenants=tenants,
        patterns=patterns,
        caps=caps,
        packs=packs,
        actors=actors,
        atypes=atypes,
        risks=risks,
        requires=requires,
        labels=labels,
        mcp=mcp,
        exact=exact,
        snapshot=snapshot,
    )


def interner_overflowed(interner: Interner, values) -> bool:
    return any(interner.lookup(v) < 0 for v in values if v.strip())


@dataclass
class JobBatch:
    any_bits: torch.Tensor  # int64 [J, 7, W]
    all_bits: torch.Tensor  # int64 [J, 2, W]
    secrets: torch.Tensor  # uint8 [J]
    mcp_bits: torch.Tensor  # int64 [J, 4, W]
    mcp_used: torch.Tensor  # uint8 [J]

    @property
    def n_jobs(self) -> int:
        return int(self.any_bits.shape[0])

    def to(self, device) -> "JobBatch":
        return JobBatch(
            self.any_bits.to(device),
            self.all_bits.to(device),
            self.secrets.to(device),
            self.mcp_bits.to(device),
            self.mcp_used.to(device),
        )


class JobEncoder:
    """Encodes PolicyInputs into device bitmasks against a compiled policy's
    vocabularies. Distinct topics' glob-pattern membership is computed once
    on the host (exact path.Match) and cached."""

    def __init__(self, compiled: CompiledPolicy):
        self.c = compiled
        self._topic_cache: Dict[str, List[int]] = {}

    def topic_mask(self, topic: str) -> List[int]:
        m = self._topic_cache.get(topic)
        if m is None:
            ids = [i for i, p in enumerate(self.c.patterns) if topic_matches(p, topic)]
            m = bit_mask(ids, self.c.words)
            self._topic_cache[topic] = m
        return m

    def encode(self, inputs: Sequence[pol.PolicyInput]) -> JobBatch:
        c = self.c
        W = c.words
        J = len(inputs)
        any_bits = torch.zeros((J, N_ANY, W), dtype=torch.int64)
        all_bits = torch.zeros((J, N_ALL, W), dtype=torch.int64)
        secrets = torch.zeros((J,), dtype=torch.uint8)
        mcp_bits = torch.zeros((J, 4, W), dtype=torch.int64)
        mcp_used = torch.zeros((J,), dtype=torch.uint8)
        for j, inp in enumerate(inputs):
            _set(any_bits, j, DIM_TENANT, single_bit(c.tenants.lookup(inp.tenant), W))
            _set(any_bits, j, DIM_TOPIC, self.topic_mask(inp.topic))
            _set(any_bits, j, DIM_CAP, single_bit(c.caps.lookup(inp.capability), W))
            _set(any_bits, j, DIM_PACK, single_bit(c.packs.lookup(inp.pack_id), W))
            _set(any_bits, j, DIM_ACTOR, single_bit(c.actors.lookup(inp.actor_id), W))
            _set(any_bits, j, DIM_ATYPE, single_bit(c.atypes.lookup(inp.actor_type), W))
            _set(any_bits, j, DIM_RISK, bit_mask([c.risks.lookup(t) for t in inp.risk_tags], W))
            _set(all_bits, j, ALL_REQUIRES, bit_mask([c.requires.lookup(t) for t in inp.requires], W))
            _set(all_bits, j, ALL_LABELS, bit_mask([c.labels.lookup(f"{k}={v}") for k, v in inp.labels.items()], W))
            secrets[j] = 1 if inp.secrets_present else 0
            used = inp.mcp.used()
            mcp_used[j] = 1 if used else 0
            if used:
                vals = (inp.mcp.server, inp.mcp.tool, inp.mcp.resource, inp.mcp.action)
                for f in range(4):
                    _set3(mcp_bits, j, f, single_bit(c.mcp.lookup(vals[f]), W))
        return JobBatch(any_bits, all_bits, secrets, mcp_bits, mcp_used)


def _set(t: torch.Tensor, j: int, d: int, mask: List[int]):
    for w, v in enumerate(mask):
        t[j, d, w] = v


def _set3(t: torch.Tensor, j: int, f: int, mask: List[int]):
    for w, v in enumerate(mask):
        t[j, f, w] = v


def first_match_reference(c: CompiledPolicy, jobs: JobBatch) -> torch.Tensor:
    """Pure-torch oracle for the K1 kernel: returns int32 [J] first-match rule
    index (-1 = no match / default allow). Used for CPU tests and as the GPU
    numerics reference."""
    R = c.n_rules
    J = jobs.n_jobs
    if R == 0 or J == 0:
        return torch.full((J,), -1, dtype=torch.int32)
    # [J, R, D, W]
    any_r = c.any_masks.unsqueeze(0)  # [1, R, 7, W]
    any_j = jobs.any_bits.unsqueeze(1)  # [J, 1, 7, W]
    rule_unconstrained = (any_r == 0).all(dim=-1)  # [1, R, 7]
    intersects = (any_r & any_j).ne(0).any(dim=-1)  # [J, R, 7]
    any_ok = (rule_unconstrained | intersects).all(dim=-1)  # [J, R]

    all_r = c.all_masks.unsqueeze(0)
    all_j = jobs.all_bits.unsqueeze(1)
    all_ok = ((all_r & ~all_j) == 0).all(dim=-1).all(dim=-1)  # [J, R]

    sec_r = c.secrets.unsqueeze(0)  # [1, R]
    sec_j = jobs.secrets.unsqueeze(1).to(torch.int8)  # [J, 1]
    sec_ok = (sec_r < 0) | (sec_r == sec_j)

    # MCP: pass when request unused or rule has no mcp; else per-field:
    # !(deny & bit) && (allow == 0 || (allow & bit))
    allow = c.mcp_masks[:, :, 0, :].unsqueeze(0)  # [1, R, 4, W]
    deny = c.mcp_masks[:, :, 1, :].unsqueeze(0)
    bits = jobs.mcp_bits.unsqueeze(1)  # [J, 1, 4, W]
    denied = (deny & bits).ne(0).any(dim=-1)  # [J, R, 4]
    allow_empty = (allow == 0).all(dim=-1)
    allowed = (allow & bits).ne(0).any(dim=-1)
    field_ok = (~denied) & (allow_empty | allowed)
    mcp_ok_fields = field_ok.all(dim=-1)  # [J, R]
    used = jobs.mcp_used.unsqueeze(1).bool()  # [J, 1]
    mcp_ok = (~used) | mcp_ok_fields

    match = any_ok & all_ok & sec_ok & mcp_ok  # [J, R]
    idx = torch.arange(R, dtype=torch.int32)
    big = torch.iinfo(torch.int32).max
    cand = torch.where(match, idx.unsqueeze(0).expand(J, R), torch.full((J, R), big, dtype=torch.int32))
    first = cand.min(dim=1).values
    return torch.where(first == big, torch.full_like(first, -1), first)
