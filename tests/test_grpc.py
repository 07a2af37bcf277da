"""gRPC surface tests: CordumApi.SubmitJob/GetJobStatus, SafetyKernel
Evaluate/ListSnapshots, ContextEngine BuildWindow/UpdateMemory — raw-bytes
grpc channel against the in-process server (contract: api.proto,
context.proto, kernel.go:106-127)."""
import json

import grpc
import pytest

from cordum_amd.gateway.grpc_api import (
    BuildWindowRequest,
    BuildWindowResponse,
    GetJobStatusRequest,
    GetJobStatusResponse,
    ListSnapshotsResponse,
    SubmitJobRequest,
    SubmitJobResponse,
    UpdateMemoryRequest,
    serve_grpc,
)
from cordum_amd.protocol.capv2 import DecisionType, PolicyCheckRequest, PolicyCheckResponse
from cordum_amd.runtime.node import Node
from cordum_amd.scheduler import PoolProfile, PoolRouting
from cordum_amd.utils.clock import ManualClock


@pytest.fixture(scope="module")
def grpc_node():
    routing = PoolRouting(topics={"job.default": ["default"]}, pools={"default": PoolProfile()})
    node = Node(clock=ManualClock(), routing=routing).start()
    node.add_worker("w1")
    server = serve_grpc(node, "127.0.0.1:0", api_keys=["gkey"])
    port = server.add_insecure_port("127.0.0.1:0")
    # the port bound in serve_grpc is unknown; rebind a fresh server instead
    server.stop(0)
    from cordum_amd.gateway.grpc_api import make_grpc_server

    server = make_grpc_server(node, api_keys=["gkey"])
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    chan = grpc.insecure_channel(f"127.0.0.1:{port}")
    yield node, chan
    server.stop(0)


def call(chan, method, req_bytes, api_key="gkey"):
    fn = chan.unary_unary(method, request_serializer=lambda b: b, response_deserializer=lambda b: b)
    return fn(req_bytes, metadata=(("x-api-key", api_key),))


def test_submit_and_status(grpc_node):
    node, chan = grpc_node
    req = SubmitJobRequest(prompt="hello grpc", topic="job.default", org_id="default")
    raw = call(chan, "/cordum.v1.CordumApi/SubmitJob", req.encode())
    resp = SubmitJobResponse.decode(raw)
    assert resp.job_id and resp.trace_id
    raw = call(chan, "/cordum.v1.CordumApi/GetJobStatus",
               GetJobStatusRequest(job_id=resp.job_id).encode())
    st = GetJobStatusResponse.decode(raw)
    assert st.status == "SUCCEEDED"
    assert st.result_ptr.startswith("redis://res:")


def test_submit_requires_auth(grpc_node):
    node, chan = grpc_node
    with pytest.raises(grpc.RpcError) as e:
        call(chan, "/cordum.v1.CordumApi/SubmitJob",
             SubmitJobRequest(prompt="x").encode(), api_key="wrong")
    assert e.value.code() == grpc.StatusCode.UNAUTHENTICATED


def test_safety_kernel_grpc(grpc_node):
    node, chan = grpc_node
    req = PolicyCheckRequest(job_id="j", topic="job.default", tenant="default")
    raw = call(chan, "/cordum.v1.SafetyKernel/Evaluate", req.encode())
    resp = PolicyCheckResponse.decode(raw)
    assert resp.decision == DecisionType.ALLOW
    raw = call(chan, "/cordum.v1.SafetyKernel/ListSnapshots", b"")
    snaps = ListSnapshotsResponse.decode(raw)
    assert node.safety_kernel.snapshot in snaps.snapshots


def test_context_engine_grpc(grpc_node):
    node, chan = grpc_node
    up = UpdateMemoryRequest(memory_id="m1",
                             logical_payload=json.dumps({"prompt": "hi"}).encode(),
                             model_response=b"there", mode=2)
    call(chan, "/cordum.v1.ContextEngine/UpdateMemory", up.encode())
    bw = BuildWindowRequest(memory_id="m1", mode=2,
                            logical_payload=json.dumps({"prompt": "again"}).encode())
    raw = call(chan, "/cordum.v1.ContextEngine/BuildWindow", bw.encode())
    resp = BuildWindowResponse.decode(raw)
    # decode repeated messages manually (field 1)
    from cordum_amd.protocol.capv2 import _dec_varint

    msgs = []
    i, n = 0, len(raw)
    while i < n:
        key, i = _dec_varint(raw, i)
        if key & 7 == 2:
            ln, i = _dec_varint(raw, i)
            if key >> 3 == 1:
                from cordum_amd.gateway.grpc_api import ModelMessagePb

                msgs.append(ModelMessagePb.decode(raw[i:i + ln]))
            i += ln
        else:
            _, i = _dec_varint(raw, i)
    contents = [m.content for m in msgs]
    assert contents == ["hi", "there", "again"]


def test_submit_tenant_from_auth_context(grpc_node):
    """A SubmitJob that omits org_id lands in the authenticated tenant
    (gateway.go:4149-4159), not the global default."""
    node, stub = grpc_node
    from cordum_amd.gateway.grpc_api import SubmitJobRequest, SubmitJobResponse, make_grpc_server
    import grpc as _grpc

    server = make_grpc_server(node, api_keys=["gkey"], auth_tenant="acme")
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    try:
        ch = _grpc.insecure_channel(f"127.0.0.1:{port}")
        call = ch.unary_unary(
            "/cordum.v1.CordumApi/SubmitJob",
            request_serializer=lambda b: b, response_deserializer=lambda b: b)
        resp = SubmitJobResponse.decode(call(
            SubmitJobRequest(prompt="hi", topic="job.echo").encode(),
            metadata=(("x-api-key", "gkey"),)))
        node.drain()
        meta = node.job_store.get_job_meta(resp.job_id)
        assert meta.get("tenant") == "acme"
    finally:
        server.stop(0)
