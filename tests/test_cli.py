"""cordumctl CLI: parse + client-path tests against a live in-process node
(mirrors the reference's cordumctl_smoke.sh, but as a unit suite)."""
import json
import threading

import pytest

from cordum_amd.cli.cordumctl import main as ctl_main


@pytest.fixture(scope="module")
def server():
    import uvicorn

    from cordum_amd.gateway.app import create_app
    from cordum_amd.gateway.auth import BasicAuthProvider
    from cordum_amd.runtime.node import Node
    from cordum_amd.scheduler import PoolProfile, PoolRouting

    node = Node(routing=PoolRouting(topics={"job.default": ["default"],
                                            "job.echo": ["default"]},
                                    pools={"default": PoolProfile()})).start()
    node.add_worker("w1", topics=["job.default", "job.echo"])
    app = create_app(node, auth=BasicAuthProvider(api_keys=["k"]))
    cfg = uvicorn.Config(app, host="127.0.0.1", port=18231, log_level="error")
    srv = uvicorn.Server(cfg)
    t = threading.Thread(target=srv.run, daemon=True)
    t.start()
    import time
    for _ in range(100):
        if srv.started:
            break
        time.sleep(0.05)
    yield "http://127.0.0.1:18231"
    srv.should_exit = True
    t.join(timeout=5)


def run_ctl(server, *argv, capsys=None):
    rc = ctl_main(["--server", server, "--api-key", "k", *argv])
    out = capsys.readouterr().out if capsys else ""
    return rc, out


def test_status(server, capsys):
    rc, out = run_ctl(server, "status", capsys=capsys)
    assert rc in (0, None)
    assert "ok" in out or "uptime" in out


def test_job_submit_and_status(server, capsys):
    rc, out = run_ctl(server, "job", "submit", "--topic", "job.echo",
                      "--prompt", "hello", capsys=capsys)
    assert rc in (0, None)
    doc = json.loads(out)
    jid = doc.get("job_id") or doc.get("jobId")
    assert jid
    rc, out = run_ctl(server, "job", "status", jid, capsys=capsys)
    assert rc in (0, None)
    assert jid in out


def test_workflow_create_and_run(server, capsys):
    wf = {"id": "cliwf", "steps": {"a": {"type": "worker", "topic": "job.echo",
                                         "input": {"prompt": "x"}}}}
    import tempfile, yaml, os
    with tempfile.NamedTemporaryFile("w", suffix=".yaml", delete=False) as f:
        yaml.safe_dump(wf, f)
        path = f.name
    try:
        rc, out = run_ctl(server, "workflow", "create", "--file", path, capsys=capsys)
        assert rc in (0, None)
        rc, out = run_ctl(server, "run", "start", "cliwf", capsys=capsys)
        assert rc in (0, None)
        run_id = json.loads(out).get("run_id") or json.loads(out).get("id")
        assert run_id
        rc, out = run_ctl(server, "run", "timeline", run_id, capsys=capsys)
        assert rc in (0, None)
    finally:
        os.unlink(path)


# --- SDK client against the live server --------------------------------------


@pytest.fixture(scope="module")
def sdk(server):
    from cordum_amd.sdk.client import Client

    return Client(base_url=server, api_key="k",
                  principal_id="tester", role="admin")


def test_sdk_job_lifecycle(sdk):
    doc = sdk.submit_job("hello sdk", topic="job.echo")
    jid = doc.get("job_id") or doc.get("jobId")
    assert jid
    job = sdk.get_job(jid)
    assert job.get("id") == jid  # detail envelope uses `id` (gateway.go:1011)
    listing = sdk.list_jobs(limit=5)
    assert "items" in listing


def test_sdk_artifacts_and_memory(sdk):
    ptr = sdk.artifacts_put(b"sdk-bytes")["ptr"]
    assert ptr.startswith("redis://art:")
    got = sdk.memory(ptr)
    assert got is not None


def test_sdk_policy_evaluate(sdk):
    res = sdk.policy_evaluate(tenant="default", topic="job.echo")
    dec = res.get("decision") or res.get("result", {}).get("decision")
    assert dec is not None


def test_sdk_status_and_workers(sdk):
    assert sdk.status().get("status") in ("ok", "degraded")
    assert "workers" in sdk.workers() or "items" in sdk.workers()


def test_pack_scaffold_installs(tmp_path):
    """`cordumctl pack scaffold` output must install cleanly end-to-end."""
    from fastapi.testclient import TestClient

    from cordum_amd.cli.cordumctl import build_pack_archive, cmd_pack_scaffold
    from cordum_amd.gateway import create_app
    from cordum_amd.runtime.node import Node

    cmd_pack_scaffold(tmp_path / "p", "scafpack")
    node = Node().start()
    client = TestClient(create_app(node))
    blob = build_pack_archive(tmp_path / "p")
    r = client.post("/api/v1/packs/install", content=blob,
                    headers={"X-Principal-Id": "p", "X-Principal-Role": "admin"})
    assert r.status_code == 200, r.text
    assert r.json()["status"] == "ACTIVE"
    r = client.get("/api/v1/workflows/scafpack.echo", headers={"X-Principal-Id": "p"})
    assert r.status_code == 200


def test_reference_format_pack_installs(tmp_path):
    """The REFERENCE's manifest dialect (metadata.id, topics[].name,
    {id,path} resources, file-based overlays with strategy keys — the exact
    shape of examples/hello-pack/pack.yaml + pack_create.go templates) must
    install unmodified."""
    import io
    import tarfile

    from fastapi.testclient import TestClient

    from cordum_amd.gateway import create_app
    from cordum_amd.runtime.node import Node

    files = {
        "pack.yaml": """apiVersion: cordum.io/v1alpha1
kind: Pack
metadata:
  id: ref-pack
  version: 0.1.0
  title: Ref Pack
  description: reference-dialect pack.
compatibility:
  protocolVersion: 1
  minCoreVersion: 0.6.0
topics:
  - name: job.ref.echo
    capability: ref.echo
resources:
  schemas:
    - id: ref-pack/EchoInput
      path: schemas/EchoInput.json
  workflows:
    - id: ref-pack.echo
      path: workflows/echo.yaml
overlays:
  config:
    - name: pools
      scope: system
      key: default
      strategy: json_merge_patch
      path: overlays/pools.patch.yaml
  policy:
    - name: safety
      strategy: bundle_fragment
      path: overlays/policy.fragment.yaml
""",
        "schemas/EchoInput.json": '{"type": "object", "required": ["message"]}',
        "workflows/echo.yaml": """id: ref-pack.echo
name: Ref Echo
org_id: default
steps:
  echo:
    type: worker
    topic: job.ref.echo
    input:
      message: "${input.message}"
""",
        "overlays/pools.patch.yaml": """pools:
  topics:
    job.ref.echo: [default]
  pools:
    default: {}
""",
        "overlays/policy.fragment.yaml": """version: ref-v1
rules:
  - id: ref-allow
    decision: allow
    match:
      topics: ["job.ref.echo"]
""",
    }
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        for name, content in files.items():
            info = tarfile.TarInfo(name)
            data = content.encode()
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))
    node = Node().start()
    client = TestClient(create_app(node))
    r = client.post("/api/v1/packs/install", content=buf.getvalue(),
                    headers={"X-Principal-Id": "p", "X-Principal-Role": "admin"})
    assert r.status_code == 200, r.text
    plan = r.json()["plan"]
    assert plan["schemas"][0]["id"] == "ref-pack/EchoInput"
    assert plan["workflows"][0]["id"] == "ref-pack.echo"
    assert plan["config_overlays"] == 1 and plan["policy_overlays"] == 1
    # routing overlay landed: job.ref.echo routes to the default pool
    r = client.get("/api/v1/workflows/ref-pack.echo", headers={"X-Principal-Id": "p"})
    assert r.status_code == 200
    # policy fragment landed under the pack-scoped bundle id
    bundles = client.get("/api/v1/policy/bundles",
                         headers={"X-Principal-Id": "p"}).json()["items"]
    assert any(b["id"] == "ref-pack/safety" for b in bundles)
