"""Black-box test of the server main: launch `cordumctl serve` as a real
subprocess, drive it with the SDK client and the platform smoke script
(reference seam: tools/scripts/platform_smoke.sh against a compose stack)."""
import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

import pytest

pytestmark = pytest.mark.timeout(120)

REPO = Path(__file__).resolve().parent.parent


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(scope="module")
def server():
    port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    proc = subprocess.Popen(
        [sys.executable, "-m", "cordum_amd.cli.cordumctl", "serve",
         "--port", str(port), "--workers", "2"],
        cwd=str(REPO), env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    )
    base = f"http://127.0.0.1:{port}"
    from cordum_amd.sdk.client import Client

    client = Client(base_url=base, role="admin", principal_id="e2e")
    for _ in range(100):
        try:
            client.status()
            break
        except Exception:
            if proc.poll() is not None:
                out = proc.stdout.read().decode()
                raise RuntimeError(f"server died: {out[-2000:]}")
            time.sleep(0.2)
    else:
        proc.kill()
        raise RuntimeError("server did not come up")
    yield base, client
    proc.send_signal(signal.SIGTERM)
    try:
        proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        proc.kill()


def test_sdk_against_live_server(server):
    base, client = server
    assert client.status()["status"] == "ok"
    # submit a job through the real HTTP stack
    r = client.submit_job("hello live", topic="job.default")
    job_id = r["job_id"]
    for _ in range(50):
        d = client.get_job(job_id)
        if d["state"] == "SUCCEEDED":
            break
        time.sleep(0.1)
    assert d["state"] == "SUCCEEDED"
    assert d["result"] == {"prompt": "hello live"}
    # workflow end to end
    client.create_workflow({
        "id": "live-wf",
        "steps": {"echo": {"type": "worker", "topic": "job.default",
                           "input": {"m": "${input.m}"}}},
    })
    run_id = client.start_run("live-wf", {"m": "x"})["run_id"]
    for _ in range(50):
        run = client.get_run(run_id)
        if run["status"] in ("succeeded", "failed"):
            break
        time.sleep(0.1)
    assert run["status"] == "succeeded"


def test_platform_smoke_script(server):
    base, _ = server
    env = dict(os.environ, CORDUM_SERVER=base, CORDUM_ROLE="admin", PYTHONPATH=str(REPO))
    res = subprocess.run(
        ["bash", str(REPO / "tools" / "scripts" / "platform_smoke.sh")],
        cwd=str(REPO), env=env, capture_output=True, text=True, timeout=60,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    assert "smoke OK" in res.stdout


def test_crash_recovery_with_checkpoint(tmp_path):
    """Failure recovery: SIGKILL the node mid-flight; restart from the
    checkpoint dir; completed state survives and WAL'd submissions that were
    never processed are replayed (reference contract: crash-safe state via
    synchronous Redis + JetStream redelivery)."""
    ckdir = str(tmp_path / "ck")
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def start():
        port = free_port()
        proc = subprocess.Popen(
            [sys.executable, "-m", "cordum_amd.cli.cordumctl", "serve",
             "--port", str(port), "--workers", "1",
             "--checkpoint-dir", ckdir, "--checkpoint-interval", "0.5"],
            cwd=str(REPO), env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
        from cordum_amd.sdk.client import Client

        client = Client(base_url=f"http://127.0.0.1:{port}", role="admin", principal_id="chaos")
        for _ in range(100):
            try:
                client.status()
                return proc, client
            except Exception:
                time.sleep(0.2)
        proc.kill()
        raise RuntimeError("no server")

    proc, client = start()
    job_id = client.submit_job("survive me", topic="job.default")["job_id"]
    for _ in range(50):
        if client.get_job(job_id)["state"] == "SUCCEEDED":
            break
        time.sleep(0.1)
    assert client.get_job(job_id)["state"] == "SUCCEEDED"
    time.sleep(1.5)  # let a checkpoint land
    proc.kill()  # SIGKILL: no graceful shutdown
    proc.wait(timeout=10)

    proc2, client2 = start()
    try:
        d = client2.get_job(job_id)
        assert d["state"] == "SUCCEEDED"
        assert d["result"] == {"prompt": "survive me"}
    finally:
        proc2.send_signal(signal.SIGTERM)
        try:
            proc2.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc2.kill()


def test_serve_device_dispatch_mode():
    """`cordumctl serve` with CORDUM_DISPATCH=device runs the batched K1/K2
    engine behind the gateway (ref backend on CPU; the GPU box runs the same
    path on the HIP extension — tests/test_gpu_serve.py)."""
    port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO), CORDUM_DISPATCH="device")
    proc = subprocess.Popen(
        [sys.executable, "-m", "cordum_amd.cli.cordumctl", "serve",
         "--port", str(port), "--workers", "2"],
        cwd=str(REPO), env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    )
    try:
        from cordum_amd.sdk.client import Client

        client = Client(base_url=f"http://127.0.0.1:{port}", role="admin",
                        principal_id="e2e")
        for _ in range(100):
            try:
                client.status()
                break
            except Exception:
                if proc.poll() is not None:
                    raise RuntimeError(proc.stdout.read().decode()[-2000:])
                time.sleep(0.2)
        out = client.submit_job(topic="job.default", prompt="via device engine")
        jid = out["job_id"]
        for _ in range(100):
            d = client.get_job(jid)
            if d["state"] in ("SUCCEEDED", "FAILED", "DENIED"):
                break
            time.sleep(0.1)
        assert d["state"] == "SUCCEEDED", d
        assert d["result"] is not None
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_external_worker_over_tcp_bridge(tmp_path):
    """The external-worker seam end-to-end: a worker in a SEPARATE process
    attaches over the TCP bus bridge (CAP v2 BusPacket frames), receives a
    dispatched job on its exclusive topic, fetches the context over HTTP,
    stores the result as an artifact, and publishes the JobResult back over
    the wire — the reference's sdk/runtime worker flow without NATS."""
    import json

    port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO), CORDUM_BRIDGE_PORT="0")
    proc = subprocess.Popen(
        [sys.executable, "-m", "cordum_amd.cli.cordumctl", "serve",
         "--port", str(port), "--workers", "1"],
        cwd=str(REPO), env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    worker_proc = None
    try:
        # the server prints the auto-assigned bridge port
        bridge_port = None
        for _ in range(200):
            line = proc.stdout.readline()
            if "bus bridge listening" in line:
                bridge_port = int(line.rsplit(":", 1)[1])
                break
            if proc.poll() is not None:
                raise RuntimeError("server died")
        assert bridge_port

        from cordum_amd.sdk.client import Client

        client = Client(base_url=f"http://127.0.0.1:{port}", role="admin",
                        principal_id="e2e")
        for _ in range(100):
            try:
                client.status()
                break
            except Exception:
                time.sleep(0.2)

        # route an exclusive topic to the default pool so only the external
        # worker can serve it
        client._req("POST", "/api/v1/config", json={
            "scope": "system", "id": "default", "merge": True,
            "config": {"pools": {"topics": {"job.external": ["default"]}}},
        })

        worker_code = f"""
import sys, json
sys.path.insert(0, {str(REPO)!r})
from cordum_amd.sdk.remote_worker import RemoteWorker

def handler(req, ctx):
    data = json.loads(ctx) if ctx else {{}}
    return json.dumps({{"echoed": data.get("prompt"), "via": "tcp-bridge"}}).encode()

w = RemoteWorker("ext-1", handler=handler, topics=["job.external"],
                 bridge_port={bridge_port},
                 api_base="http://127.0.0.1:{port}")
print("worker up", flush=True)
w.run_forever()
"""
        worker_proc = subprocess.Popen([sys.executable, "-c", worker_code],
                                       cwd=str(REPO), env=env,
                                       stdout=subprocess.PIPE,
                                       stderr=subprocess.STDOUT, text=True)
        assert "worker up" in worker_proc.stdout.readline()
        time.sleep(0.5)  # heartbeat lands, registry knows the worker

        out = client.submit_job(topic="job.external", prompt="over the wire")
        jid = out["job_id"]
        d = None
        for _ in range(150):
            d = client.get_job(jid)
            if d["state"] in ("SUCCEEDED", "FAILED", "DENIED"):
                break
            time.sleep(0.1)
        assert d and d["state"] == "SUCCEEDED", d
        assert d["result"]["echoed"] == "over the wire"
        assert d["result"]["via"] == "tcp-bridge"
    finally:
        if worker_proc is not None:
            worker_proc.kill()
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_crash_restart_recovers_state(tmp_path):
    """Kill -9 the serving process and restart it on the same state dir:
    checkpoint + WAL replay must restore completed job state and re-drive
    unfinished submissions (reference: crash-safe via synchronous Redis +
    JetStream redelivery; here: snapshot + fsync'd WAL through idempotent
    handlers)."""
    port = free_port()
    state = str(tmp_path / "state")
    # device dispatch engine (ref backend on CPU): the WAL replay must
    # re-drive submissions through the batched path too
    env = dict(os.environ, PYTHONPATH=str(REPO), CORDUM_DISPATCH="device")

    def start():
        return subprocess.Popen(
            [sys.executable, "-m", "cordum_amd.cli.cordumctl", "serve",
             "--port", str(port), "--workers", "2",
             "--checkpoint-dir", state, "--checkpoint-interval", "1"],
            cwd=str(REPO), env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        )

    from cordum_amd.sdk.client import Client

    client = Client(base_url=f"http://127.0.0.1:{port}", role="admin",
                    principal_id="e2e")

    def wait_up(proc):
        for _ in range(150):
            try:
                client.status()
                return
            except Exception:
                if proc.poll() is not None:
                    raise RuntimeError(proc.stdout.read().decode()[-1500:])
                time.sleep(0.2)
        raise RuntimeError("server did not come up")

    proc = start()
    try:
        wait_up(proc)
        done = client.submit_job(topic="job.default", prompt="will finish")
        for _ in range(100):
            d = client.get_job(done["job_id"])
            if d["state"] == "SUCCEEDED":
                break
            time.sleep(0.1)
        assert d["state"] == "SUCCEEDED"
        # a job on an unserved topic parks PENDING/DISPATCH-pending
        stuck = client.submit_job(topic="job.nobody", prompt="survives crash",
                                  idempotency_key="crash-1")
        time.sleep(2.5)  # let a checkpoint + WAL fsync land
        proc.kill()      # SIGKILL: no graceful shutdown
        proc.wait(timeout=10)

        proc = start()
        wait_up(proc)
        d = client.get_job(done["job_id"])
        assert d["state"] == "SUCCEEDED"  # terminal state survived
        d2 = client.get_job(stuck["job_id"])
        assert d2["state"] not in ("", "UNSPECIFIED"), d2  # submission survived
        # idempotency survived the crash too: same key dedups to the old job
        again = client.submit_job(topic="job.nobody", prompt="retry after crash",
                                  idempotency_key="crash-1")
        assert again["job_id"] == stuck["job_id"]
        assert again.get("deduplicated") is True
    finally:
        proc.kill()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            pass
