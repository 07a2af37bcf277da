"""Auth: pluggable provider + OSS basic API-key implementation.

Oracle: gateway/auth_provider.go:8-47 (AuthProvider interface),
gateway/basic_auth.go:20-233 (flat API-key allowlist from CORDUM_API_KEYS
json/csv or CORDUM_API_KEY/API_KEY, single tenant TENANT_ID, principal via
X-Principal-Id, role via X-Principal-Role with secops/operator -> admin,
WS key via `Sec-WebSocket-Protocol: cordum-api-key, <base64url>`), plus the
enterprise extension seams (RouteRegistrar/PublicPathProvider/AuditExporter/
LicenseInfoProvider — gateway/extensions.go:9-38).
"""
from __future__ import annotations

import base64
import json
import os
from dataclasses import dataclass
from typing import Dict, List, Optional, Protocol


@dataclass
class Principal:
    id: str = ""
    role: str = "user"
    tenant: str = "default"

    @property
    def is_admin(self) -> bool:
        return self.role == "admin"


class AuthProvider(Protocol):
    def authenticate(self, headers: Dict[str, str]) -> Optional[Principal]: ...

    def require_role(self, principal: Principal, role: str) -> bool: ...

    def resolve_tenant(self, principal: Principal, requested: str) -> str: ...


def _normalize_role(raw: str) -> str:
    r = (raw or "").strip().lower()
    if r in ("admin", "secops", "operator"):
        return "admin"
    return r or "user"


class BasicAuthProvider:
    """OSS provider: flat API-key allowlist, single tenant."""

    def __init__(self, api_keys: Optional[List[str]] = None, tenant: str = ""):
        if api_keys is None:
            api_keys = self._keys_from_env()
        self.api_keys = set(k for k in api_keys if k)
        self.tenant = tenant or os.environ.get("TENANT_ID", "default")

    @staticmethod
    def _keys_from_env() -> List[str]:
        raw = os.environ.get("CORDUM_API_KEYS", "")
        keys: List[str] = []
        if raw:
            try:
                parsed = json.loads(raw)
                if isinstance(parsed, list):
                    keys = [str(k) for k in parsed]
            except ValueError:
                keys = [k.strip() for k in raw.split(",") if k.strip()]
        for env in ("CORDUM_API_KEY", "API_KEY"):
            v = os.environ.get(env, "")
            if v:
                keys.append(v)
        return keys

    def _key_ok(self, key: str) -> bool:
        if not self.api_keys:
            return True  # no keys configured -> open (dev mode), as in OSS basic auth
        return key in self.api_keys

    def authenticate(self, headers: Dict[str, str]) -> Optional[Principal]:
        h = {k.lower(): v for k, v in headers.items()}
        key = h.get("x-api-key", "")
        if not key:
            # WS handshake: Sec-WebSocket-Protocol: cordum-api-key, <base64url>
            proto = h.get("sec-websocket-protocol", "")
            if proto:
                parts = [p.strip() for p in proto.split(",")]
                if len(parts) >= 2 and parts[0] == "cordum-api-key":
                    try:
                        pad = "=" * (-len(parts[1]) % 4)
                        key = base64.urlsafe_b64decode(parts[1] + pad).decode("utf-8")
                    except Exception:
                        key = ""
        if not self._key_ok(key):
            return None
        return Principal(
            id=h.get("x-principal-id", ""),
            role=_normalize_role(h.get("x-principal-role", "")),
            tenant=self.tenant,
        )

    def require_role(self, principal: Principal, role: str) -> bool:
        # OSS RequireRole is a no-op unless role is admin-ish; enterprise enforces
        if role == "admin":
            return principal.is_admin
        return True

    def resolve_tenant(self, principal: Principal, requested: str) -> str:
        return requested or principal.tenant

    def require_tenant_access(self, principal: Principal, tenant: str) -> bool:
        # OSS single-tenant: everything in the configured tenant is accessible
        return True


class TokenBucket:
    """API rate limit (API_RATE_LIMIT_RPS/BURST, gateway.go:117-165)."""

    def __init__(self, rps: float, burst: int, clock=None):
        from ..utils.clock import SYSTEM_CLOCK

        self.rps = rps
        self.burst = burst
        self.clock = clock or SYSTEM_CLOCK
        self._tokens = float(burst)
        self._last = self.clock.now()

    def allow(self) -> bool:
        if self.rps <= 0:
            return True
        now = self.clock.now()
        self._tokens = min(self.burst, self._tokens + (now - self._last) * self.rps)
        self._last = now
        if self._tokens >= 1:
            self._tokens -= 1
            return True
        return False
