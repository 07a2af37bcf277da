"""Full gateway route parity: every HTTP route the reference mux
registers (gateway.go:701-805, extracted verbatim) must be served here.
The dashboard suite covers the dashboard's call sites; this pins the whole
compat surface including routes only the CLI/SDK use."""
import re

import pytest

from cordum_amd.gateway import create_app
from cordum_amd.runtime.node import Node

REFERENCE_ROUTES = [
    ("DELETE", "/api/v1/dlq/{job_id}"),
    ("DELETE", "/api/v1/schemas/{id}"),
    ("DELETE", "/api/v1/workflow-runs/{id}"),
    ("DELETE", "/api/v1/workflows/{id}"),
    ("GET", "/api/v1/approvals"),
    ("GET", "/api/v1/artifacts/{ptr}"),
    ("GET", "/api/v1/config"),
    ("GET", "/api/v1/config/effective"),
    ("GET", "/api/v1/dlq"),
    ("GET", "/api/v1/dlq/page"),
    ("GET", "/api/v1/jobs"),
    ("GET", "/api/v1/jobs/{id}"),
    ("GET", "/api/v1/jobs/{id}/decisions"),
    ("GET", "/api/v1/locks"),
    ("GET", "/api/v1/marketplace/packs"),
    ("GET", "/api/v1/memory"),
    ("GET", "/api/v1/packs"),
    ("GET", "/api/v1/packs/{id}"),
    ("GET", "/api/v1/policy/audit"),
    ("GET", "/api/v1/policy/bundles"),
    ("GET", "/api/v1/policy/bundles/snapshots"),
    ("GET", "/api/v1/policy/bundles/snapshots/{id}"),
    ("GET", "/api/v1/policy/bundles/{id}"),
    ("GET", "/api/v1/policy/rules"),
    ("GET", "/api/v1/policy/snapshots"),
    ("GET", "/api/v1/schemas"),
    ("GET", "/api/v1/schemas/{id}"),
    ("GET", "/api/v1/status"),
    ("GET", "/api/v1/traces/{id}"),
    ("GET", "/api/v1/workers"),
    ("GET", "/api/v1/workflow-runs"),
    ("GET", "/api/v1/workflow-runs/{id}"),
    ("GET", "/api/v1/workflow-runs/{id}/timeline"),
    ("GET", "/api/v1/workflows"),
    ("GET", "/api/v1/workflows/{id}"),
    ("GET", "/api/v1/workflows/{id}/runs"),
    ("POST", "/api/v1/approvals/{job_id}/approve"),
    ("POST", "/api/v1/approvals/{job_id}/reject"),
    ("POST", "/api/v1/artifacts"),
    ("POST", "/api/v1/config"),
    ("POST", "/api/v1/dlq/{job_id}/retry"),
    ("POST", "/api/v1/jobs"),
    ("POST", "/api/v1/jobs/{id}/cancel"),
    ("POST", "/api/v1/jobs/{id}/remediate"),
    ("POST", "/api/v1/locks/acquire"),
    ("POST", "/api/v1/locks/release"),
    ("POST", "/api/v1/locks/renew"),
    ("POST", "/api/v1/marketplace/install"),
    ("POST", "/api/v1/packs/install"),
    ("POST", "/api/v1/packs/{id}/uninstall"),
    ("POST", "/api/v1/packs/{id}/verify"),
    ("POST", "/api/v1/policy/bundles/snapshots"),
    ("POST", "/api/v1/policy/bundles/{id}/simulate"),
    ("POST", "/api/v1/policy/evaluate"),
    ("POST", "/api/v1/policy/explain"),
    ("POST", "/api/v1/policy/publish"),
    ("POST", "/api/v1/policy/rollback"),
    ("POST", "/api/v1/policy/simulate"),
    ("POST", "/api/v1/schemas"),
    ("POST", "/api/v1/workflow-runs/{id}/rerun"),
    ("POST", "/api/v1/workflows"),
    ("POST", "/api/v1/workflows/{id}/runs"),
    ("POST", "/api/v1/workflows/{id}/runs/{run_id}/cancel"),
    ("POST", "/api/v1/workflows/{id}/runs/{run_id}/steps/{step_id}/approve"),
    ("PUT", "/api/v1/policy/bundles/{id}"),
]


@pytest.fixture(scope="module")
def route_table():
    app = create_app(Node().start())

    def walk(routes, prefix=""):
        for r in routes:
            if type(r).__name__ == "_IncludedRouter":
                yield from walk(r.original_router.routes,
                                prefix + (r.include_context.prefix or ""))
                continue
            for m in (getattr(r, "methods", None) or set()):
                yield m, re.sub(r"\{[^}]+\}", "{}", prefix + getattr(r, "path", ""))

    return set(walk(app.routes))


@pytest.mark.parametrize("method,path", REFERENCE_ROUTES,
                         ids=[f"{m} {p}" for m, p in REFERENCE_ROUTES])
def test_reference_route_served(route_table, method, path):
    norm = re.sub(r"\{[^}]+\}", "{}", path)
    assert (method, norm) in route_table, f"reference serves {method} {path}"
