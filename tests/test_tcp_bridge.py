"""TCP bus bridge unit tests: framing, wildcard subscriptions, queue-group
load balancing with in-process subscribers, and publish-through semantics
(the external-worker seam; full process-level e2e in test_serve_e2e.py)."""
import threading
import time

import pytest

from cordum_amd.bus.tcp_bridge import BridgeClient, BusBridgeServer
from cordum_amd.protocol.capv2 import BusPacket, JobRequest, JobResult, JobStatus
from cordum_amd.runtime.node import Node

pytestmark = pytest.mark.timeout(60)


@pytest.fixture
def node():
    return Node().start()


@pytest.fixture
def bridge(node):
    srv = BusBridgeServer(node, port=0).start()
    yield srv
    srv.stop()


def collect(client, out, n):
    def loop():
        for _ in range(n):
            msg = client.next_message()
            if msg is None:
                return
            out.append(msg)

    t = threading.Thread(target=loop, daemon=True)
    t.start()
    return t


def test_subscribe_wildcard_and_roundtrip(node, bridge):
    c = BridgeClient(port=bridge.port)
    c.subscribe("sys.job.>")
    time.sleep(0.1)
    got = []
    t = collect(c, got, 1)
    node.bus.publish("sys.job.result", BusPacket(
        trace_id="t1", protocol_version=1,
        job_result=JobResult(job_id="j1", status=JobStatus.SUCCEEDED)))
    t.join(timeout=5)
    assert len(got) == 1
    subject, pkt = got[0]
    assert subject == "sys.job.result"
    assert pkt.job_result.job_id == "j1"
    assert pkt.trace_id == "t1"
    c.close()


def test_publish_through_reaches_node_bus(node, bridge):
    seen = []
    node.bus.subscribe("job.bridge", lambda s, p: seen.append(p.job_request.job_id))
    c = BridgeClient(port=bridge.port)
    c.publish("job.bridge", BusPacket(
        protocol_version=1, job_request=JobRequest(job_id="from-wire",
                                                   topic="job.bridge")))
    for _ in range(50):
        if seen:
            break
        time.sleep(0.05)
    assert seen == ["from-wire"]
    c.close()


def test_queue_group_load_balances_across_wire_clients(node, bridge):
    a = BridgeClient(port=bridge.port)
    b = BridgeClient(port=bridge.port)
    a.subscribe("job.pool", queue_group="g")
    b.subscribe("job.pool", queue_group="g")
    time.sleep(0.1)
    got_a, got_b = [], []
    ta = collect(a, got_a, 10)
    tb = collect(b, got_b, 10)
    for i in range(10):
        node.bus.publish("job.pool", BusPacket(
            protocol_version=1,
            job_request=JobRequest(job_id=f"j{i}", topic="job.pool")))
    deadline = time.time() + 5
    while time.time() < deadline and len(got_a) + len(got_b) < 10:
        time.sleep(0.05)
    assert len(got_a) + len(got_b) == 10  # each job delivered exactly once
    assert got_a and got_b               # and both members shared the load
    a.close()
    b.close()


def test_disconnected_client_unsubscribes(node, bridge):
    c = BridgeClient(port=bridge.port)
    c.subscribe("sys.job.>")
    time.sleep(0.1)
    c.close()
    time.sleep(0.2)
    # a publish after disconnect must not error or leak to the dead socket
    node.bus.publish("sys.job.result", BusPacket(
        protocol_version=1,
        job_result=JobResult(job_id="gone", status=JobStatus.SUCCEEDED)))
    assert all(not conn.alive for conn in bridge._conns)


def test_dead_client_does_not_fail_publisher(node, bridge):
    """A wire client that dies mid-delivery must NAK (durable subjects
    redeliver to survivors) instead of raising into the publisher."""
    from cordum_amd.protocol import subjects as subj

    dead = BridgeClient(port=bridge.port)
    dead.subscribe("job.pool2", queue_group="g2")
    live = BridgeClient(port=bridge.port)
    live.subscribe("job.pool2", queue_group="g2")
    time.sleep(0.1)
    # kill the first client's socket without unsubscribing
    dead.sock.close()
    time.sleep(0.1)
    got = []
    t = collect(live, got, 4)
    for i in range(4):
        # durable subject: delivery to the dead member NAKs and the pump
        # redelivers; publish must never raise
        node.bus.publish("job.pool2", BusPacket(
            protocol_version=1,
            job_request=JobRequest(job_id=f"d{i}", topic="job.pool2")))
    deadline = time.time() + 5
    while time.time() < deadline and len(got) < 4:
        node.bus.pump()
        time.sleep(0.05)
    assert len(got) == 4  # every message landed on the survivor
    live.close()


def test_frame_roundtrip_fuzz(node, bridge):
    """Random subjects + random CAP payloads through real sockets: frames
    must round-trip exactly (length-prefix framing, UTF-8 subjects, binary
    payload bytes)."""
    import random

    from cordum_amd.protocol.capv2 import Heartbeat, JobRequest

    rng = random.Random(9)
    c = BridgeClient(port=bridge.port)
    c.subscribe("fuzz.>")
    time.sleep(0.1)
    sent = []
    for i in range(40):
        depth = rng.randint(1, 4)
        subject = "fuzz." + ".".join(
            "".join(rng.choice("abcxyz09_-") for _ in range(rng.randint(1, 8)))
            for _ in range(depth))
        pkt = BusPacket(
            trace_id="".join(rng.choice("0123456789abcdef") for _ in range(16)),
            protocol_version=1,
            job_request=JobRequest(
                job_id=f"fz{i}", topic=subject,
                labels={f"k{j}": "".join(rng.choice("µ€漢x") for _ in range(3))
                        for j in range(rng.randint(0, 3))},
            ) if rng.random() < 0.7 else None,
            heartbeat=None if rng.random() < 0.7 else Heartbeat(worker_id=f"w{i}"),
        )
        sent.append((subject, pkt.encode()))
        node.bus.publish(subject, pkt)
    got = []
    t = collect(c, got, len(sent))
    t.join(timeout=10)
    assert len(got) == len(sent)
    for (subject, wire), (gs, gp) in zip(sent, got):
        assert gs == subject
        assert gp.encode() == wire  # byte-exact through the socket
    c.close()


def test_remote_and_inprocess_workers_share_pool_group(node, bridge):
    """A pool-topic publish must reach exactly ONE member of the pool even
    when the pool mixes an in-process worker and a wire-attached worker
    (same queue-group convention; a divergent group name double-delivered)."""
    from cordum_amd.sdk.remote_worker import RemoteWorker

    hits = []
    node.add_worker("inproc-1", handler=lambda req, ctx: hits.append(req.job_id) or b"{}",
                    topics=["job.mixed"])

    ext_hits = []

    class _NoApi:
        def memory(self, ptr):
            return b"{}"

        def artifacts_put(self, b):
            return {"ptr": ""}

    w = RemoteWorker.__new__(RemoteWorker)
    # wire only the bus piece (no HTTP server in this unit test)
    from cordum_amd.bus.tcp_bridge import BridgeClient
    import threading

    w.worker_id = "ext-1"
    w.pool = "default"
    w.topics = ["job.mixed"]
    w.bus = BridgeClient(port=bridge.port)
    w.bus.subscribe("job.mixed", queue_group="pool.default")
    time.sleep(0.1)

    def drainer():
        while True:
            msg = w.bus.next_message()
            if msg is None:
                return
            _, pkt = msg
            if pkt.job_request is not None:
                ext_hits.append(pkt.job_request.job_id)

    threading.Thread(target=drainer, daemon=True).start()

    for i in range(12):
        node.bus.publish("job.mixed", BusPacket(
            protocol_version=1,
            job_request=JobRequest(job_id=f"m{i}", topic="job.mixed")))
    node.drain()
    deadline = time.time() + 5
    while time.time() < deadline and len(hits) + len(ext_hits) < 12:
        time.sleep(0.05)
    assert len(hits) + len(ext_hits) == 12  # exactly-once per publish
    assert hits and ext_hits                # both members shared the load
    w.bus.close()
