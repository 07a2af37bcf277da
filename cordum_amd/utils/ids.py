"""ID generation (UUIDv4 strings, matching the reference's google/uuid usage)."""
from __future__ import annotations

import uuid


def new_id() -> str:
    return str(uuid.uuid4())


def new_trace_id() -> str:
    return str(uuid.uuid4())


def short_id(n: int = 8) -> str:
    return uuid.uuid4().hex[:n]
