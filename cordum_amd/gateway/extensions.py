"""Enterprise extension seams (oracle: gateway/extensions.go:9-38,
license_info.go:3-21): the OSS gateway exposes the same hook surface the
reference does, so closed-source add-ons can register routes, open public
paths, export audit events, and report license info without patching the
gateway.

Pass implementations via `create_app(..., extensions=[...])`; each object
is probed for the protocol methods it implements (an object may implement
several)."""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Protocol, runtime_checkable


@dataclass
class AuditEvent:
    """One exported audit record (gateway/extensions.go AuditEvent)."""

    ts: float = field(default_factory=time.time)
    subject: str = ""
    actor: str = ""
    action: str = ""
    resource: str = ""
    detail: Dict[str, Any] = field(default_factory=dict)


@runtime_checkable
class RouteRegistrar(Protocol):
    """Registers extra routes on the FastAPI app at create_app time."""

    def register_routes(self, app) -> None: ...


@runtime_checkable
class PublicPathProvider(Protocol):
    """Paths (exact or prefix ending '/') served WITHOUT authentication."""

    def public_paths(self) -> List[str]: ...


@runtime_checkable
class AuditExporter(Protocol):
    """Receives every sys.audit.> event the node publishes."""

    def export_audit(self, event: AuditEvent) -> None: ...


@runtime_checkable
class LicenseInfoProvider(Protocol):
    """Overrides the license block in GET /api/v1/status."""

    def license_info(self) -> Dict[str, Any]: ...


def path_is_public(path: str, public: List[str]) -> bool:
    for p in public:
        if p.endswith("/"):
            if path.startswith(p):
                return True
        elif path == p:
            return True
    return False
