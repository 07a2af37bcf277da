"""MFMA variant of the K1 policy kernel: rule×job matching as int8 matrix
products on the CDNA4 matrix cores.

Each set-match dimension is lowered to one-hot int8 vectors over the interned
vocabulary (vocab 64 = one `v_mfma_i32_16x16x64_i8` per dimension per
16-job × 16-rule tile):

  dot(job_onehot, rule_onehot) = |job_set ∩ rule_set|
  any-of dim passes  <=>  card(rule)=0 or dot>0
  all-of dim passes  <=>  dot == card(rule)

This module prepacks jobs/rules into the MFMA fragment layout
(A: lane l -> A[row=l&15][k=(l>>4)*16..+15]; B: lane l ->
B[k=(l>>4)*16..+15][col=l&15]; C/D: col=l&15, row=(l>>4)*4+reg — guide §3)
and drives the policy_first_match_mfma kernel.

WHY THIS IS NOT THE DEFAULT PATH (measured, see profiles/):
one-hot rules cost 9×64 = 576 B/rule of stream traffic versus 136 B/rule for
the bitset rows of the default kernel, and the MFMA tiles cannot early-exit
at the first matching rule. The matrix cores execute the arithmetic
essentially for free — the workload is rule-stream-bound, so the 4.2×
representation inflation decides it. Kept as a correct, tested alternative
and as the measured justification for the bitset design.
Scope: the 7 any-of + 2 all-of set dimensions + secrets (the synthetic and
benchmark policies); MCP-constrained rules use the default bitset kernel.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import torch

from .policy_compile import CompiledPolicy, JobBatch

N_DIMS = 9  # 7 any-of + 2 all-of
VOCAB = 64


def _bits_to_onehot(words: torch.Tensor) -> torch.Tensor:
    """int64 [N, D, 1] bit words -> uint8 one-hot [N, D, 64]."""
    n, d, w = words.shape
    assert w == 1, "MFMA variant supports 1-word (vocab<=64) policies"
    shifts = torch.arange(64, dtype=torch.int64)
    return ((words >> shifts.view(1, 1, 64)) & 1).to(torch.int8)


def _pack_a(onehot: torch.Tensor) -> torch.Tensor:
    """jobs one-hot [J, 9, 64] -> A fragments [Jt, 9, 64 lanes, 16] int8.

    lane l of tile jt holds A[row = l&15][k = (l>>4)*16 + t]."""
    J = onehot.shape[0]
    Jt = (J + 15) // 16
    padded = torch.zeros(Jt * 16, N_DIMS, VOCAB, dtype=torch.int8)
    padded[:J] = onehot
    a = padded.view(Jt, 16, N_DIMS, 4, 16)  # [jt, row, d, kblock, t]
    # lane = kblock*16 + row -> index [jt, d, lane, t]
    a = a.permute(0, 2, 3, 1, 4).reshape(Jt, N_DIMS, 64, 16)
    return a.contiguous()


def _pack_b(onehot: torch.Tensor) -> torch.Tensor:
    """rules one-hot [R, 9, 64] -> B fragments [Rt, 9, 64 lanes, 16] int8.

    lane l of tile rt holds B[k = (l>>4)*16 + t][col = l&15]."""
    R = onehot.shape[0]
    Rt = (R + 15) // 16
    padded = torch.zeros(Rt * 16, N_DIMS, VOCAB, dtype=torch.int8)
    padded[:R] = onehot
    b = padded.view(Rt, 16, N_DIMS, 4, 16)  # [rt, col, d, kblock, t]
    b = b.permute(0, 2, 3, 1, 4)  # [rt, d, kblock, col, t]
    # lane = kblock*16 + col holds bytes t -> index [rt, d, lane= kb*16+col, t]
    b = b.reshape(Rt, N_DIMS, 64, 16)
    return b.contiguous()


@dataclass
class MfmaPolicy:
    b_pack: torch.Tensor  # [Rt, 9, 64, 16] int8
    cards: torch.Tensor  # [Rt*16, 9] int32 (padded rules have impossible cards)
    secrets: torch.Tensor  # [Rt*16] int8
    tile_dims: torch.Tensor  # [Rt] int32 bitmask: dims any rule in the tile uses
    n_rules: int

    def to(self, device):
        return MfmaPolicy(self.b_pack.to(device), self.cards.to(device),
                          self.secrets.to(device), self.tile_dims.to(device),
                          self.n_rules)


def pack_policy_mfma(c: CompiledPolicy) -> MfmaPolicy:
    assert c.words == 1, "MFMA variant needs vocab <= 64"
    R = c.n_rules
    rule_bits = torch.cat([c.any_masks, c.all_masks], dim=1)  # [R, 9, 1]
    onehot = _bits_to_onehot(rule_bits)
    b_pack = _pack_b(onehot)
    Rt = b_pack.shape[0]
    cards = torch.zeros(Rt * 16, N_DIMS, dtype=torch.int32)
    cards[:R] = onehot.sum(dim=2).to(torch.int32)
    # padding rules must never match: impossible all-of card on dim 7
    if Rt * 16 > R:
        cards[R:, 7] = 127
    secrets = torch.full((Rt * 16,), -1, dtype=torch.int8)
    secrets[:R] = c.secrets
    # per-tile union of constrained dims: a dim with card==0 for every rule
    # in the tile auto-passes, so the kernel skips its MFMA + B load
    used = (cards.view(Rt, 16, N_DIMS) != 0).any(dim=1)  # [Rt, 9]
    tile_dims = (used.to(torch.int32) << torch.arange(N_DIMS, dtype=torch.int32)).sum(
        dim=1).to(torch.int32)
    return MfmaPolicy(b_pack, cards, secrets, tile_dims, R)


def pack_jobs_mfma(jobs: JobBatch) -> Tuple[torch.Tensor, torch.Tensor]:
    job_bits = torch.cat([jobs.any_bits, jobs.all_bits], dim=1)  # [J, 9, 1]
    onehot = _bits_to_onehot(job_bits)
    return _pack_a(onehot), jobs.secrets


def first_match_mfma(ext, mp: MfmaPolicy, a_pack: torch.Tensor,
                     job_secrets: torch.Tensor, n_jobs: int) -> torch.Tensor:
    return ext.policy_first_match_mfma(
        a_pack, mp.b_pack, mp.cards, mp.secrets, job_secrets, mp.tile_dims,
        int(n_jobs), int(mp.n_rules),
    )
