"""Canonical JSON: stable serialization for config snapshot hashing.

Oracle: core/configsvc/canonical_json.go — sorted keys, no insignificant
whitespace, UTF-8. Used for EffectiveSnapshot.hash (configsvc/service.go:105-149)
and scheduler overlay hashing (cmd/cordum-scheduler/config_overlay.go).
"""
from __future__ import annotations

import hashlib
import json
from typing import Any


def canonical_json(value: Any) -> str:
    return json.dumps(value, sort_keys=True, separators=(",", ":"), ensure_ascii=False)


def canonical_json_hash(value: Any) -> str:
    return hashlib.sha256(canonical_json(value).encode("utf-8")).hexdigest()
