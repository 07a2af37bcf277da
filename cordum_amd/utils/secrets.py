"""Secret-reference detection/redaction.

Oracle: core/infra/secrets/secrets.go:8-100 — detect `secret://` refs
recursively in any JSON value; the gateway sets risk tag `secrets` and label
`secrets_present` at submit (gateway.go:1820-1827).
"""
from __future__ import annotations

from typing import Any

SECRET_SCHEME = "secret://"
REDACTED = "secret://redacted"


def contains_secret_refs(value: Any) -> bool:
    if isinstance(value, str):
        return SECRET_SCHEME in value
    if isinstance(value, dict):
        return any(contains_secret_refs(v) for v in value.values()) or any(
            contains_secret_refs(k) for k in value.keys()
        )
    if isinstance(value, (list, tuple)):
        return any(contains_secret_refs(v) for v in value)
    return False


def redact_secret_refs(value: Any) -> Any:
    if isinstance(value, str):
        return REDACTED if SECRET_SCHEME in value else value
    if isinstance(value, dict):
        return {k: redact_secret_refs(v) for k, v in value.items()}
    if isinstance(value, list):
        return [redact_secret_refs(v) for v in value]
    return value
