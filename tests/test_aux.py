"""Auxiliary subsystems: context engine, WAL/checkpoint, config loaders,
metrics, logging, secrets, CLI scaffolding."""
import io
import json
import sys

import pytest

from cordum_amd.config import NodeConfig, load_pools, load_timeouts
from cordum_amd.protocol import JobState
from cordum_amd.protocol.capv2 import JobRequest
from cordum_amd.runtime.context_engine import ContextEngine
from cordum_amd.runtime.node import Node
from cordum_amd.scheduler import PoolProfile, PoolRouting
from cordum_amd.store import MemoryStore
from cordum_amd.store.wal import Checkpointer
from cordum_amd.utils import logging as clog
from cordum_amd.utils.clock import ManualClock
from cordum_amd.utils.metrics import Metrics
from cordum_amd.utils.secrets import contains_secret_refs, redact_secret_refs


# --- context engine (oracle context/engine/service.go) -----------------------


@pytest.fixture
def ctx_engine():
    mem = MemoryStore(clock=ManualClock())
    return ContextEngine(mem), mem


def test_raw_window(ctx_engine):
    eng, _ = ctx_engine
    w = eng.build_window("m1", mode="raw", logical_payload=b'{"prompt": "hello"}')
    assert [m.to_dict() for m in w.messages] == [{"role": "user", "content": "hello"}]
    assert w.output_tokens == 1024


def test_chat_window_history_and_trim(ctx_engine):
    eng, _ = ctx_engine
    for i in range(25):
        eng.update_memory("m2", json.dumps({"prompt": f"q{i}"}).encode(), f"a{i}".encode())
    w = eng.build_window("m2", mode="chat", logical_payload=b'{"prompt": "latest"}')
    # history trimmed to 20 events + the new prompt
    assert len(w.messages) == 21
    assert w.messages[-1].content == "latest"
    # token budget trim drops oldest
    w2 = eng.build_window("m2", mode="chat", logical_payload=b'{"prompt": "latest"}',
                          max_input_tokens=0)
    assert len(w2.messages) == 1 and w2.messages[0].content == "latest"


def test_rag_window_chunks_filtered_by_path(ctx_engine):
    eng, _ = ctx_engine
    eng.put_summary("m3", "repo summary")
    eng.put_chunk("m3", 0, "chunk a", file_path="a.py")
    eng.put_chunk("m3", 1, "chunk b", file_path="b.py")
    w = eng.build_window("m3", mode="rag",
                         logical_payload=json.dumps({"prompt": "q", "file_path": "a.py"}).encode())
    contents = [m.content for m in w.messages]
    assert contents == ["repo summary", "chunk a", "q"]


# --- WAL / checkpoint ----------------------------------------------------------


def make_node():
    clock = ManualClock()
    routing = PoolRouting(topics={"job.default": ["default"]}, pools={"default": PoolProfile()})
    n = Node(clock=clock, routing=routing).start()
    n.add_worker("w1")
    return n


def test_checkpoint_restore_roundtrip(tmp_path):
    n = make_node()
    req = JobRequest(job_id="cj1", topic="job.default", tenant_id="default")
    n.submit_job(req, context=b'{"x": 1}')
    n.drain()
    assert n.job_store.get_state("cj1") == JobState.SUCCEEDED
    n.configsvc.set("org", "acme", {"retry": {"max": 9}})
    ck = Checkpointer(n, str(tmp_path))
    ck.checkpoint()

    n2 = make_node()
    ck2 = Checkpointer(n2, str(tmp_path))
    assert ck2.restore()
    assert n2.job_store.get_state("cj1") == JobState.SUCCEEDED
    assert n2.configsvc.get("org", "acme") == {"retry": {"max": 9}}
    assert n2.memory.get("ctx:cj1") == b'{"x": 1}'


def test_wal_replay_redrives_unfinished_jobs(tmp_path):
    n = make_node()
    ck = Checkpointer(n, str(tmp_path))
    ck.checkpoint()  # empty baseline
    req = JobRequest(job_id="wj1", topic="job.default", tenant_id="default")
    ck.wal_append(req, "tr1")
    # crash before processing: new node restores checkpoint + replays WAL
    n2 = make_node()
    ck2 = Checkpointer(n2, str(tmp_path))
    ck2.restore()
    assert n2.job_store.get_state("wj1") == JobState.UNSPECIFIED
    assert ck2.replay_wal() == 1
    assert n2.job_store.get_state("wj1") == JobState.SUCCEEDED


# --- config loaders --------------------------------------------------------------


def test_load_pools_and_timeouts(tmp_path):
    p = tmp_path / "pools.yaml"
    p.write_text("topics:\n  job.x: default\n  job.multi: [a, b]\npools:\n  default:\n    requires: [gpu]\n")
    routing = load_pools(str(p))
    assert routing.topics["job.x"] == ["default"]
    assert routing.topics["job.multi"] == ["a", "b"]
    assert routing.pools["default"].requires == ["gpu"]

    t = tmp_path / "timeouts.yaml"
    t.write_text("reconciler:\n  dispatch_timeout_seconds: 60\n  running_timeout_seconds: 120\n")
    to = load_timeouts(str(t))
    assert to.dispatch_timeout_s == 60 and to.running_timeout_s == 120
    assert load_timeouts(str(tmp_path / "missing.yaml")).dispatch_timeout_s == 300


def test_default_config_files_load():
    cfg = NodeConfig.from_env()
    routing = load_pools(cfg.pool_config_path)
    assert "job.default" in routing.topics
    from cordum_amd.config import load_safety_yaml
    from cordum_amd.safety import parse_safety_policy

    policy = parse_safety_policy(load_safety_yaml(cfg.safety_policy_path))
    assert policy is not None and "default" in policy.tenants


# --- metrics / logging -------------------------------------------------------------


def test_metrics_exposition():
    m = Metrics()
    m.inc_received("job.x")
    m.inc_dispatched("job.x")
    m.inc_completed("job.x", "SUCCEEDED")
    m.inc_safety_denied("job.y")
    text = m.exposition().decode()
    assert "cordum_scheduler_jobs_received_total" in text
    assert "cordum_scheduler_jobs_completed_total" in text


def test_logging_formats():
    buf = io.StringIO()
    clog.configure(level="info", json_format=False, out=buf)
    clog.info("scheduler", "job dispatched", job_id="j1", topic="job.x")
    clog.debug("scheduler", "hidden")
    line = buf.getvalue()
    assert "job dispatched" in line and "job_id=j1" in line and "hidden" not in line
    buf2 = io.StringIO()
    clog.configure(level="info", json_format=True, out=buf2)
    clog.error("gateway", "boom", trace_id="t1")
    rec = json.loads(buf2.getvalue())
    assert rec["level"] == "error" and rec["trace_id"] == "t1"
    clog.configure(level="info", json_format=False, out=sys.stderr)


# --- secrets -------------------------------------------------------------------------


def test_secrets_detection_and_redaction():
    v = {"a": ["x", {"b": "use secret://v/k"}], "c": 1}
    assert contains_secret_refs(v)
    red = redact_secret_refs(v)
    assert red["a"][1]["b"] == "secret://redacted"
    assert not contains_secret_refs({"a": "plain"})


# --- CLI -----------------------------------------------------------------------------


def test_cli_init_and_pack_create(tmp_path):
    from cordum_amd.cli.cordumctl import main

    assert main(["init", str(tmp_path / "proj")]) == 0
    assert (tmp_path / "proj" / "config" / "pools.yaml").exists()
    assert (tmp_path / "proj" / "workflows" / "hello.json").exists()
    # pack create from the hello-pack example
    out = tmp_path / "hello.tgz"
    assert main(["pack", "create", "--dir", "examples/hello-pack", "-o", str(out)]) == 0
    assert out.exists() and out.stat().st_size > 0
    # and that archive installs cleanly through the gateway
    from fastapi.testclient import TestClient

    from cordum_amd.gateway import BasicAuthProvider, create_app

    n = make_node()
    app = create_app(n, auth=BasicAuthProvider(api_keys=["k"]))
    c = TestClient(app)
    c.headers.update({"X-API-Key": "k", "X-Principal-Role": "admin"})
    r = c.post("/api/v1/packs/install", content=out.read_bytes())
    assert r.status_code == 200, r.text
    assert r.json()["pack_id"] == "hello-pack"


# --- ed25519 policy signatures -------------------------------------------------


def test_signed_policy_load(tmp_path):
    from cordum_amd.config import load_safety_yaml
    from cordum_amd.utils import ed25519

    policy = tmp_path / "safety.yaml"
    policy.write_text("version: signed\ntenants: {}\n")
    sk = b"\x01" * 32
    pk = ed25519.public_key(sk)
    sig = ed25519.sign(sk, policy.read_bytes())
    # valid signature
    text = load_safety_yaml(str(policy), public_key=pk.hex(), signature=sig.hex())
    assert "signed" in text
    # tampered content fails
    policy.write_text("version: tampered\ntenants: {}\n")
    with pytest.raises(ValueError):
        load_safety_yaml(str(policy), public_key=pk.hex(), signature=sig.hex())
    # no key configured -> no verification
    assert "tampered" in load_safety_yaml(str(policy))
