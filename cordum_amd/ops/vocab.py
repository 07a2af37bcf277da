"""String interning + bitmask vocabularies for the device policy/scoring path.

Every string-valued match dimension (tenants, topic patterns, capabilities,
risk tags, requires, pack ids, actor ids, label k=v pairs, MCP fields) is
interned into a small integer vocabulary and represented as bits in int64
words. `words` is the number of 64-bit words per dimension (vocab capacity =
64*words); policies/jobs whose vocabulary overflows the capacity are handled
by the host evaluator (the compiler reports inexact), so the device fast path
never silently diverges from the oracle.

Matching semantics follow safety_policy.go:294-345: case-insensitive,
trimmed, empty string never matches — so interning keys are
`value.strip().lower()` and empty values get no bit.
"""
from __future__ import annotations

from typing import Dict, Iterable, List


class Interner:
    def __init__(self, capacity: int, casefold: bool = True):
        # labels are matched case-SENSITIVELY (labelsMatch, safety_policy.go:330),
        # every other dimension case-insensitively (containsString :296).
        self.capacity = capacity
        self.casefold = casefold
        self._ids: Dict[str, int] = {}
        self.overflow = False

    def _key(self, value: str) -> str:
        v = value.strip()
        return v.lower() if self.casefold else v

    def intern(self, value: str) -> int:
        """Returns bit index, or -1 for empty/overflow."""
        key = self._key(value)
        if not key:
            return -1
        idx = self._ids.get(key)
        if idx is not None:
            return idx
        if len(self._ids) >= self.capacity:
            self.overflow = True
            return -1
        idx = len(self._ids)
        self._ids[key] = idx
        return idx

    def lookup(self, value: str) -> int:
        return self._ids.get(self._key(value), -1)

    def __len__(self):
        return len(self._ids)

    def items(self):
        return self._ids.items()


def bit_mask(indices: Iterable[int], words: int) -> List[int]:
    """Pack bit indices into `words` int64 words (python ints, two's complement)."""
    out = [0] * words
    for i in indices:
        if i < 0:
            continue
        w, b = divmod(i, 64)
        if w < words:
            out[w] |= 1 << b
    # convert to signed int64 range for torch
    return [(v - (1 << 64)) if v >= (1 << 63) else v for v in out]


def single_bit(idx: int, words: int) -> List[int]:
    return bit_mask([idx] if idx >= 0 else [], words)
