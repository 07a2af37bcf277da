"""Gateway HTTP tests (reference seam: httptest suites — gateway_test.go,
jobs_http_test.go, approvals_test.go, packs_*_test.go, policy_bundles_test.go)."""
import io
import json
import tarfile

import pytest
from fastapi.testclient import TestClient

from cordum_amd.gateway import BasicAuthProvider, create_app
from cordum_amd.protocol import JobState
from cordum_amd.runtime.node import Node
from cordum_amd.scheduler import PoolProfile, PoolRouting
from cordum_amd.utils.clock import ManualClock

POLICY = """
version: v1
default_tenant: default
rules:
  - id: approve-risky
    decision: require_approval
    reason: risky
    match: {risk_tags: [risky]}
  - id: deny-bad
    decision: deny
    reason: blocked
    match: {topics: ["job.bad"]}
"""


@pytest.fixture
def node():
    clock = ManualClock()
    routing = PoolRouting(
        topics={"job.default": ["default"], "job.echo": ["default"]},
        pools={"default": PoolProfile()},
    )
    n = Node(clock=clock, routing=routing, policy_yaml=POLICY).start()
    n.add_worker("w1", topics=["job.default", "job.echo"])
    return n


@pytest.fixture
def client(node):
    auth = BasicAuthProvider(api_keys=["test-key"], tenant="default")
    app = create_app(node, auth=auth)
    c = TestClient(app)
    c.headers.update({"X-API-Key": "test-key", "X-Principal-Id": "tester",
                      "X-Principal-Role": "admin"})
    return c


def test_auth_required(node):
    app = create_app(node, auth=BasicAuthProvider(api_keys=["k1"]))
    c = TestClient(app)
    assert c.get("/api/v1/status").status_code == 401
    assert c.get("/api/v1/status", headers={"X-API-Key": "k1"}).status_code == 200


def test_submit_and_get_job(client, node):
    r = client.post("/api/v1/jobs", json={"prompt": "hello world", "topic": "job.echo"})
    assert r.status_code == 200, r.text
    job_id = r.json()["job_id"]
    trace_id = r.json()["trace_id"]
    r = client.get(f"/api/v1/jobs/{job_id}")
    assert r.status_code == 200
    d = r.json()
    assert d["state"] == "SUCCEEDED"
    assert d["topic"] == "job.echo"
    assert d["result"] == {"prompt": "hello world"}
    assert d["context"]["prompt"] == "hello world"
    assert d["safety_decision"] == "allow"
    assert d["trace_id"] == trace_id
    # trace endpoint
    r = client.get(f"/api/v1/traces/{trace_id}")
    assert r.json()["jobs"][0]["id"] == job_id


def test_submit_validation(client):
    assert client.post("/api/v1/jobs", json={}).status_code == 400  # no prompt
    assert client.post("/api/v1/jobs", json={"prompt": "x", "topic": "sys.bad"}).status_code == 400
    assert client.post("/api/v1/jobs", json={"prompt": "x", "deadline_ms": -1}).status_code == 400
    assert client.post("/api/v1/jobs", json={"prompt": "x", "actor_type": "robot"}).status_code == 400
    labels = {f"k{i}": "v" for i in range(51)}
    assert client.post("/api/v1/jobs", json={"prompt": "x", "labels": labels}).status_code == 400


def test_submit_idempotency(client):
    r1 = client.post("/api/v1/jobs", json={"prompt": "a", "topic": "job.echo", "idempotency_key": "same"})
    r2 = client.post("/api/v1/jobs", json={"prompt": "a", "topic": "job.echo", "idempotency_key": "same"})
    assert r1.json()["job_id"] == r2.json()["job_id"]
    assert r2.json().get("deduplicated") is True


def test_secrets_scan_triggers_risk_tag(client, node):
    r = client.post("/api/v1/jobs", json={"prompt": "use secret://vault/key", "topic": "job.echo"})
    job_id = r.json()["job_id"]
    d = client.get(f"/api/v1/jobs/{job_id}").json()
    assert "secrets" in d["risk_tags"]


def test_job_list_filters(client, node):
    client.post("/api/v1/jobs", json={"prompt": "a", "topic": "job.echo"})
    client.post("/api/v1/jobs", json={"prompt": "b", "topic": "job.bad"})
    r = client.get("/api/v1/jobs", params={"state": "SUCCEEDED"})
    assert all(i["state"] == "SUCCEEDED" for i in r.json()["items"])
    r = client.get("/api/v1/jobs", params={"topic": "job.bad"})
    assert all(i["topic"] == "job.bad" for i in r.json()["items"])


def test_deny_path_populates_dlq(client, node):
    r = client.post("/api/v1/jobs", json={"prompt": "x", "topic": "job.bad"})
    job_id = r.json()["job_id"]
    d = client.get(f"/api/v1/jobs/{job_id}").json()
    assert d["state"] == "DENIED"
    assert d["error_code"] == "safety_denied"
    dlq = client.get("/api/v1/dlq").json()["items"]
    assert any(e["job_id"] == job_id for e in dlq)


def test_dlq_retry_endpoint(client, node):
    r = client.post("/api/v1/jobs", json={"prompt": "x", "topic": "job.bad"})
    job_id = r.json()["job_id"]
    r = client.post(f"/api/v1/dlq/{job_id}/retry")
    assert r.status_code == 200
    new_id = r.json()["job_id"]
    assert new_id.startswith(job_id + "-retry-")
    d = client.get(f"/api/v1/jobs/{new_id}").json()
    assert d["state"] == "DENIED"  # still denied by policy, but retried through the machine
    assert d["labels"]["retry_of_job"] == job_id


def test_approval_http_flow(client, node):
    r = client.post("/api/v1/jobs", json={"prompt": "x", "topic": "job.echo", "risk_tags": ["risky"]})
    job_id = r.json()["job_id"]
    assert client.get(f"/api/v1/jobs/{job_id}").json()["state"] == "APPROVAL_REQUIRED"
    items = client.get("/api/v1/approvals").json()["items"]
    assert any(i["id"] == job_id for i in items)
    r = client.post(f"/api/v1/approvals/{job_id}/approve", json={"reason": "looks fine"})
    assert r.status_code == 200, r.text
    d = client.get(f"/api/v1/jobs/{job_id}").json()
    assert d["state"] == "SUCCEEDED"
    assert d["approval_by"] == "tester"
    assert d["approval_reason"] == "looks fine"


def test_approval_requires_admin(node):
    auth = BasicAuthProvider(api_keys=["k"], tenant="default")
    app = create_app(node, auth=auth)
    c = TestClient(app)
    c.headers.update({"X-API-Key": "k", "X-Principal-Role": "user"})
    r = c.post("/api/v1/jobs", json={"prompt": "x", "topic": "job.echo", "risk_tags": ["risky"]})
    job_id = r.json()["job_id"]
    assert c.post(f"/api/v1/approvals/{job_id}/approve", json={}).status_code == 403


def test_workflow_step_approval_requires_admin(node):
    """Workflow step approval is admin-gated like job approvals
    (gateway.go:3558 requireRole "admin")."""
    auth = BasicAuthProvider(api_keys=["k"], tenant="default")
    app = create_app(node, auth=auth)
    c = TestClient(app)
    c.headers.update({"X-API-Key": "k", "X-Principal-Role": "user"})
    r = c.post("/api/v1/workflows/any/runs/any/steps/any/approve", json={})
    assert r.status_code == 403


def test_approval_reject(client, node):
    r = client.post("/api/v1/jobs", json={"prompt": "x", "topic": "job.echo", "risk_tags": ["risky"]})
    job_id = r.json()["job_id"]
    r = client.post(f"/api/v1/approvals/{job_id}/reject", json={"reason": "nope"})
    assert r.status_code == 200
    assert client.get(f"/api/v1/jobs/{job_id}").json()["state"] == "DENIED"


def test_cancel_endpoint(client, node):
    # job stuck pending (no workers for topic after removing worker)
    node.configsvc.set("system", "default", {"pools": {"topics": {"job.stuck": ["nowhere"]},
                                                       "pools": {"nowhere": {}}}})
    r = client.post("/api/v1/jobs", json={"prompt": "x", "topic": "job.stuck"})
    job_id = r.json()["job_id"]
    assert client.get(f"/api/v1/jobs/{job_id}").json()["state"] == "PENDING"
    r = client.post(f"/api/v1/jobs/{job_id}/cancel")
    assert r.status_code == 200
    assert client.get(f"/api/v1/jobs/{job_id}").json()["state"] == "CANCELLED"


def test_workflow_http_crud_and_run(client, node):
    wf = {
        "id": "hello-workflow",
        "name": "Hello",
        "input_schema": {"type": "object", "required": ["message"]},
        "steps": {"echo": {"type": "worker", "topic": "job.echo",
                           "input": {"message": "${input.message}"}}},
    }
    assert client.post("/api/v1/workflows", json=wf).status_code == 200
    assert any(w["id"] == "hello-workflow" for w in client.get("/api/v1/workflows").json()["items"])
    # schema validation rejects bad input
    r = client.post("/api/v1/workflows/hello-workflow/runs", json={"input": {}})
    assert r.status_code == 400
    r = client.post("/api/v1/workflows/hello-workflow/runs", json={"input": {"message": "hi"}})
    assert r.status_code == 200, r.text
    run_id = r.json()["run_id"]
    run = client.get(f"/api/v1/workflow-runs/{run_id}").json()
    assert run["status"] == "succeeded"
    tl = client.get(f"/api/v1/workflow-runs/{run_id}/timeline").json()["items"]
    assert any(e["type"] == "step_completed" for e in tl)
    # rerun
    r = client.post(f"/api/v1/workflow-runs/{run_id}/rerun", json={})
    assert r.status_code == 200
    assert client.get(f"/api/v1/workflow-runs/{r.json()['run_id']}").json()["status"] == "succeeded"
    # delete
    assert client.delete("/api/v1/workflows/hello-workflow").status_code == 200


def test_policy_endpoints(client, node):
    r = client.post("/api/v1/policy/evaluate", json={"topic": "job.bad", "tenant": "default"})
    assert r.json()["decision"] == "DECISION_TYPE_DENY"
    r = client.post("/api/v1/policy/explain", json={"topic": "job.echo", "tenant": "default"})
    assert "rules" in r.json()
    r = client.get("/api/v1/policy/rules")
    assert any(rule["id"] == "deny-bad" for rule in r.json()["items"])
    assert client.get("/api/v1/policy/snapshots").json()["current"]


def test_policy_bundle_studio(client, node):
    content = "version: v2\nrules:\n  - id: extra-deny\n    decision: deny\n    match: {topics: [\"job.extra\"]}\n"
    r = client.put("/api/v1/policy/bundles/team~extra", json={"content": content, "message": "add"})
    assert r.status_code == 200, r.text
    items = client.get("/api/v1/policy/bundles").json()["items"]
    assert any(b["id"] == "team/extra" for b in items)
    # the kernel picked the bundle up: evaluation now denies job.extra
    r = client.post("/api/v1/policy/evaluate", json={"topic": "job.extra", "tenant": "default"})
    assert r.json()["decision"] == "DECISION_TYPE_DENY"
    # simulate a draft without side effects
    r = client.post("/api/v1/policy/bundles/team~extra/simulate", json={
        "content": "version: v3\nrules:\n  - id: sim\n    decision: allow\n    match: {topics: [\"job.extra\"]}\n",
        "input": {"topic": "job.extra", "tenant": "default"},
    })
    assert r.json()["results"][0]["rule_id"] == "sim"
    # snapshots + publish + rollback
    snap = client.post("/api/v1/policy/bundles/snapshots", json={"message": "before"}).json()["id"]
    client.put("/api/v1/policy/bundles/team~extra", json={"content": "", "enabled": False})
    r = client.post("/api/v1/policy/rollback", json={"snapshot_id": snap})
    assert r.status_code == 200
    r = client.post("/api/v1/policy/evaluate", json={"topic": "job.extra", "tenant": "default"})
    assert r.json()["decision"] == "DECISION_TYPE_DENY"  # rollback restored the deny
    audit = client.get("/api/v1/policy/audit").json()["items"]
    assert any(e["action"] == "rollback" for e in audit)


def make_pack_tgz(manifest: dict, files: dict) -> bytes:
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        import yaml as _yaml

        def add(name, data: bytes):
            info = tarfile.TarInfo(name)
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))

        add("pack.yaml", _yaml.safe_dump(manifest).encode())
        for name, data in files.items():
            add(name, data)
    return buf.getvalue()


HELLO_PACK = {
    "apiVersion": "cordum.io/v1alpha1",
    "metadata": {"name": "hello-pack", "version": "1.0.0", "description": "echo"},
    "compatibility": {"protocolVersion": 1},
    "topics": [{"topic": "job.echo", "capability": "echo"}],
    "resources": {"schemas": ["schemas/echo.json"], "workflows": ["workflows/hello.json"]},
    "overlays": {
        "config": [{"scope": "system", "key": "default",
                    "json_merge_patch": {"pools": {"topics": {"job.echo": ["default"], "job.default": ["default"]},
                                                   "pools": {"default": {}}}}}],
        "policy": [{"id": "hello-pack/policy",
                    "bundle_fragment": "version: hp\nrules:\n  - id: hp-allow\n    decision: allow\n    match: {topics: [\"job.echo\"]}\n"}],
    },
}

HELLO_FILES = {
    "schemas/echo.json": json.dumps({"$id": "echo-input", "type": "object", "required": ["message"]}).encode(),
    "workflows/hello.json": json.dumps({
        "id": "hello-workflow", "name": "Hello",
        "steps": {"echo": {"type": "worker", "topic": "job.echo", "input": {"message": "${input.message}"}}},
    }).encode(),
}


def test_pack_install_uninstall_verify(client, node):
    blob = make_pack_tgz(HELLO_PACK, HELLO_FILES)
    r = client.post("/api/v1/packs/install", content=blob)
    assert r.status_code == 200, r.text
    assert r.json()["status"] == "ACTIVE"
    # resources landed
    assert client.get("/api/v1/schemas/echo-input").status_code == 200
    assert client.get("/api/v1/workflows/hello-workflow").status_code == 200
    packs = client.get("/api/v1/packs").json()["items"]
    assert any(p["id"] == "hello-pack" and p["status"] == "ACTIVE" for p in packs)
    # the installed workflow runs end-to-end
    r = client.post("/api/v1/workflows/hello-workflow/runs", json={"input": {"message": "from pack"}})
    assert r.status_code == 200
    assert client.get(f"/api/v1/workflow-runs/{r.json()['run_id']}").json()["status"] == "succeeded"
    # verify + uninstall (soft)
    assert client.post("/api/v1/packs/hello-pack/verify").json()["ok"] is True
    r = client.post("/api/v1/packs/hello-pack/uninstall")
    assert r.json()["status"] == "INACTIVE"
    assert client.get("/api/v1/workflows/hello-workflow").status_code == 200  # kept


def test_pack_zip_slip_rejected(client):
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        data = b"evil"
        info = tarfile.TarInfo("../../etc/passwd")
        info.size = len(data)
        tf.addfile(info, io.BytesIO(data))
    r = client.post("/api/v1/packs/install", content=buf.getvalue())
    assert r.status_code == 400


def test_locks_endpoints(client):
    assert client.post("/api/v1/locks/acquire", json={"resource": "r1", "owner": "a"}).status_code == 200
    assert client.post("/api/v1/locks/acquire", json={"resource": "r1", "owner": "b"}).status_code == 409
    assert client.get("/api/v1/locks").json()["items"][0]["resource"] == "r1"
    assert client.post("/api/v1/locks/renew", json={"resource": "r1", "owner": "a"}).status_code == 200
    assert client.post("/api/v1/locks/release", json={"resource": "r1", "owner": "a"}).json()["released"]


def test_schemas_and_config_endpoints(client):
    assert client.post("/api/v1/schemas", json={"id": "s1", "schema": {"type": "object"}}).status_code == 200
    assert client.get("/api/v1/schemas/s1").json()["schema"] == {"type": "object"}
    assert "s1" in client.get("/api/v1/schemas").json()["items"]
    assert client.delete("/api/v1/schemas/s1").status_code == 200
    r = client.post("/api/v1/config", json={"scope": "org", "id": "acme", "config": {"retry": {"max": 2}}})
    assert r.json()["revision"] == 1
    eff = client.get("/api/v1/config/effective", params={"org": "acme"}).json()
    assert eff["config"]["retry"]["max"] == 2 and eff["hash"]


def test_status_and_workers(client, node):
    s = client.get("/api/v1/status").json()
    assert s["status"] == "ok" and s["workers"] == 1
    w = client.get("/api/v1/workers").json()
    assert w["pools"]["default"]["workers"] == 1


def test_artifacts_and_memory(client, node):
    r = client.post("/api/v1/artifacts", content=b'{"x": 1}',
                    headers={"content-type": "application/json", "x-retention": "audit"})
    ptr = r.json()["ptr"]
    art_id = ptr.split("art:")[1]
    assert client.get(f"/api/v1/artifacts/{art_id}").json() == {"x": 1}
    # memory pointer reader
    node.memory.put("ctx:manual", b'{"y": 2}')
    assert client.get("/api/v1/memory", params={"ptr": "redis://ctx:manual"}).json() == {"y": 2}
    assert client.get("/api/v1/memory", params={"ptr": "redis://missing"}).status_code == 404


def test_dashboard_and_health(client):
    assert client.get("/health").json() == {"status": "ok"}
    r = client.get("/dashboard")
    assert r.status_code == 200 and "cordum-mi355x" in r.text
    assert client.get("/metrics").status_code == 200


def test_marketplace_catalog_flow(client, node):
    """packs.go:454-607: catalog registry + 30s cache + sha256-pinned
    install, with a stub fetcher standing in for the catalog HTTP server
    (like marketplace_test.go's httptest servers)."""
    import hashlib

    blob = make_pack_tgz(HELLO_PACK, HELLO_FILES)
    sha = hashlib.sha256(blob).hexdigest()
    catalog = {
        "updated_at": "2026-01-01T00:00:00Z",
        "packs": [{
            "id": "hello-pack", "version": "1.0.0", "title": "Hello",
            "url": "https://packs.example.com/hello.tgz", "sha256": sha,
        }],
    }
    fetches = []

    def fetcher(url):
        fetches.append(url)
        if url.endswith("catalog.json"):
            return json.dumps(catalog).encode()
        if url.endswith("hello.tgz"):
            return blob
        raise ValueError("unknown url " + url)

    node.marketplace_fetcher = fetcher
    node.configsvc.set("system", "pack_catalogs", {"catalogs": [
        {"id": "main", "title": "Main", "url": "https://packs.example.com/catalog.json",
         "enabled": True},
        {"id": "off", "url": "https://off.example.com/catalog.json", "enabled": False},
    ]})

    r = client.get("/api/v1/marketplace/packs")
    assert r.status_code == 200, r.text
    snap = r.json()
    assert [c["id"] for c in snap["catalogs"]] == ["main", "off"]
    assert snap["catalogs"][1]["enabled"] is False
    assert snap["items"][0]["id"] == "hello-pack"
    assert snap["items"][0]["sha256"] == sha
    # 30s cache: the second snapshot does not re-fetch
    n_fetches = len(fetches)
    r = client.get("/api/v1/marketplace/packs")
    assert r.json()["cached"] is True
    assert len(fetches) == n_fetches

    # catalog_id + pack_id install path
    r = client.post("/api/v1/marketplace/install",
                    json={"catalog_id": "main", "pack_id": "hello-pack"})
    assert r.status_code == 200, r.text
    assert r.json()["status"] == "ACTIVE"

    # URL install requires the pinned sha256 and catalog membership
    r = client.post("/api/v1/marketplace/install",
                    json={"url": "https://packs.example.com/hello.tgz"})
    assert r.status_code == 400  # sha256 required
    r = client.post("/api/v1/marketplace/install",
                    json={"url": "https://packs.example.com/hello.tgz",
                          "sha256": "deadbeef"})
    assert r.status_code == 400  # sha mismatch vs catalog pin
    r = client.post("/api/v1/marketplace/install",
                    json={"url": "https://rogue.example.com/evil.tgz",
                          "sha256": sha})
    assert r.status_code == 404  # not in any enabled catalog
    r = client.post("/api/v1/marketplace/install",
                    json={"url": "https://packs.example.com/hello.tgz",
                          "sha256": sha.upper()})
    assert r.status_code == 200  # case-insensitive digest compare

    # tampered payload: catalog pin protects the install
    def bad_fetcher(url):
        if url.endswith("catalog.json"):
            return json.dumps(catalog).encode()
        return b"tampered"

    node.marketplace_fetcher = bad_fetcher
    r = client.post("/api/v1/marketplace/install",
                    json={"catalog_id": "main", "pack_id": "hello-pack"})
    assert r.status_code == 400
    assert "sha256 mismatch" in r.text


def test_marketplace_no_fetcher_refuses(client, node):
    if hasattr(node, "marketplace_fetcher"):
        del node.marketplace_fetcher
    node.configsvc.set("system", "pack_catalogs", {"catalogs": [
        {"id": "main", "url": "https://packs.example.com/catalog.json"}]})
    snap = client.get("/api/v1/marketplace/packs").json()
    assert "error" in snap["catalogs"][0]
    assert snap["items"] == []


def test_job_detail_field_parity(client, node):
    """GET /jobs/{id} must carry the reference's full detail surface
    (gateway.go:1011-1173, ~35 fields incl. inlined context/result JSON,
    safety decision, approval audit, workflow ids)."""
    r = client.post("/api/v1/jobs", json={
        "topic": "job.default", "prompt": "detail", "capability": "echo",
        "actor_id": "a1", "actor_type": "service", "pack_id": "p1",
        "risk_tags": ["t"], "requires": ["r"], "idempotency_key": "detail-1",
        "labels": {"workflow_id": "wf9", "run_id": "r9", "step_id": "s9"},
    })
    assert r.status_code == 200, r.text
    jid = r.json()["job_id"]
    d = client.get(f"/api/v1/jobs/{jid}").json()
    for f in ("id", "state", "trace_id", "context_ptr", "context",
              "result_ptr", "topic", "tenant", "actor_id", "actor_type",
              "idempotency_key", "capability", "pack_id", "risk_tags",
              "requires", "attempts", "safety_decision", "safety_reason",
              "safety_rule_id", "safety_snapshot", "safety_constraints",
              "safety_remediations", "safety_job_hash", "approval_required",
              "approval_ref", "labels", "workflow_id", "run_id", "step_id"):
        assert f in d, f"missing job-detail field {f!r}"
    assert d["context"]["prompt"] == "detail"
    assert d["workflow_id"] == "wf9" and d["run_id"] == "r9" and d["step_id"] == "s9"
    if d["state"] == "SUCCEEDED":
        assert d["result"] is not None


def test_ws_stream_delivers_job_events(client, node):
    """/api/v1/stream delivers protojson-shaped packets for job traffic
    (gateway.go:2002-2049; the dashboard's live event feed)."""
    with client.websocket_connect(
            "/api/v1/stream",
            headers={"X-API-Key": "test-key", "X-Principal-Id": "ws-test"}) as ws:
        r = client.post("/api/v1/jobs", json={"topic": "job.echo", "prompt": "ws"})
        assert r.status_code == 200
        jid = r.json()["job_id"]
        seen_subjects = []
        payload = None
        for _ in range(20):
            msg = ws.receive_json()
            seen_subjects.append(msg["subject"])
            pkt = msg["packet"]
            if pkt.get("jobRequest", {}).get("jobId") == jid:
                payload = pkt
            if pkt.get("jobResult", {}).get("jobId") == jid:
                break
        assert payload is not None, seen_subjects
        # protojson camelCase shape
        assert payload["jobRequest"]["topic"] == "job.echo"
        assert any(s.startswith("sys.job.") for s in seen_subjects)


def test_ws_stream_subprotocol_auth(client):
    """Browser clients can't set headers on a WebSocket, so the API key rides
    the subprotocol (`Sec-WebSocket-Protocol: cordum-api-key, <b64url(key)>`,
    gateway.go:2154-2185); the server must accept AND echo the negotiated
    subprotocol or the client-side handshake validation fails."""
    import base64

    b64 = base64.urlsafe_b64encode(b"test-key").decode().rstrip("=")
    with client.websocket_connect(
            "/api/v1/stream",
            subprotocols=["cordum-api-key", b64]) as ws:
        r = client.post("/api/v1/jobs", json={"topic": "job.echo", "prompt": "sp"},
                        headers={"X-API-Key": "test-key"})
        assert r.status_code == 200
        msg = ws.receive_json()
        assert msg["subject"].startswith("sys.job.")


def test_ws_stream_rejects_bad_key(client):
    """An unauthenticated WS connect must be closed, not accepted."""
    import pytest as _pytest
    from starlette.websockets import WebSocketDisconnect

    with _pytest.raises(WebSocketDisconnect):
        with client.websocket_connect(
                "/api/v1/stream",
                headers={"X-API-Key": "wrong-key"}) as ws:
            ws.receive_json()


def test_enterprise_extension_seams(node):
    """The extension hook surface (gateway/extensions.go:9-38): route
    registrar, public paths (served without auth), audit exporter, and
    license-info provider must all take effect through create_app."""
    from fastapi.testclient import TestClient

    from cordum_amd.gateway.app import create_app
    from cordum_amd.gateway.auth import BasicAuthProvider
    from cordum_amd.protocol.capv2 import BusPacket, SystemAlert

    audited = []

    class Ext:
        def register_routes(self, app):
            @app.get("/ext/ping")
            def ping():
                return {"ext": True}

        def public_paths(self):
            return ["/api/v1/status"]

        def export_audit(self, event):
            audited.append(event)

        def license_info(self):
            return {"edition": "enterprise", "seats": 5}

    app = create_app(node, auth=BasicAuthProvider(api_keys=["k"]),
                     extensions=[Ext()])
    c = TestClient(app)
    # registrar route exists
    assert c.get("/ext/ping").json() == {"ext": True}
    # public path: no API key required; license info provider honored
    r = c.get("/api/v1/status")
    assert r.status_code == 200
    assert r.json()["license"] == {"edition": "enterprise", "seats": 5}
    # non-public path still requires auth
    assert c.get("/api/v1/jobs").status_code == 401
    # audit exporter receives sys.audit.> traffic
    node.bus.publish("sys.audit.test", BusPacket(
        trace_id="tr-1", protocol_version=1,
        alert=SystemAlert(source="unit", severity="info")))
    assert audited and audited[0].subject == "sys.audit.test"
    assert audited[0].actor == "unit" and audited[0].resource == "tr-1"
