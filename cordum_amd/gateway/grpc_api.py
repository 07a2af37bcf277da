"""gRPC surface: CordumApi, SafetyKernel and ContextEngine services.

Contracts: core/protocol/proto/v1/api.proto (CordumApi.SubmitJob/GetJobStatus
— field numbers implemented verbatim), context.proto (ContextEngine
BuildWindow/UpdateMemory — verbatim), and the SafetyKernel service
(kernel.go:106-127) carried over our CAP v2 codec (protocol/capv2.py).

Implemented with grpcio generic handlers + the in-repo protobuf codec (no
protoc available offline), so the wire format is standard protobuf over
standard gRPC framing.
"""
from __future__ import annotations

import json
from concurrent import futures
from dataclasses import dataclass, field as dc_field
from typing import Dict, List, Optional

import os

import grpc

from ..protocol.capv2 import (
    F,
    JobMetadata,
    JobRequest,
    JobPriority,
    ActorType,
    Message,
    PolicyCheckRequest,
    )


# -- api.proto messages (field numbers from core/protocol/proto/v1/api.proto) --


@dataclass(eq=False)
class SubmitJobRequest(Message):
    prompt: str = ""
    topic: str = ""
    adapter_id: str = ""
    priority: str = ""
    org_id: str = ""
    team_id: str = ""
    project_id: str = ""
    principal_id: str = ""
    idempotency_key: str = ""
    actor_id: str = ""
    actor_type: str = ""
    pack_id: str = ""
    capability: str = ""
    risk_tags: List[str] = dc_field(default_factory=list)
    requires: List[str] = dc_field(default_factory=list)
    labels: Dict[str, str] = dc_field(default_factory=dict)
    memory_id: str = ""

    FIELDS = {
        "prompt": F(1, "str"),
        "topic": F(2, "str"),
        "adapter_id": F(3, "str"),
        "priority": F(4, "str"),
        "org_id": F(5, "str"),
        "team_id": F(6, "str"),
        "project_id": F(7, "str"),
        "principal_id": F(8, "str"),
        "idempotency_key": F(9, "str"),
        "actor_id": F(10, "str"),
        "actor_type": F(11, "str"),
        "pack_id": F(12, "str"),
        "capability": F(13, "str"),
        "risk_tags": F(14, "rep_str"),
        "requires": F(15, "rep_str"),
        "labels": F(16, "map_ss"),
        "memory_id": F(17, "str"),
    }


@dataclass(eq=False)
class SubmitJobResponse(Message):
    job_id: str = ""
    trace_id: str = ""

    FIELDS = {"job_id": F(1, "str"), "trace_id": F(2, "str")}


@dataclass(eq=False)
class GetJobStatusRequest(Message):
    job_id: str = ""

    FIELDS = {"job_id": F(1, "str")}


@dataclass(eq=False)
class GetJobStatusResponse(Message):
    job_id: str = ""
    status: str = ""
    result_ptr: str = ""

    FIELDS = {"job_id": F(1, "str"), "status": F(2, "str"), "result_ptr": F(3, "str")}


# -- context.proto messages ---------------------------------------------------


@dataclass(eq=False)
class ModelMessagePb(Message):
    role: str = ""
    content: str = ""

    FIELDS = {"role": F(1, "str"), "content": F(2, "str")}


@dataclass(eq=False)
class BuildWindowRequest(Message):
    memory_id: str = ""
    mode: int = 0  # 0 unspecified, 1 raw, 2 chat, 3 rag
    model: str = ""
    logical_payload: bytes = b""
    max_input_tokens: int = 0
    max_output_tokens: int = 0

    FIELDS = {
        "memory_id": F(1, "str"),
        "mode": F(2, "int"),
        "model": F(3, "str"),
        "logical_payload": F(4, "bytes"),
        "max_input_tokens": F(5, "int"),
        "max_output_tokens": F(6, "int"),
    }


@dataclass(eq=False)
class BuildWindowResponse(Message):
    messages: List[ModelMessagePb] = dc_field(default_factory=list)
    input_tokens: int = 0
    output_tokens: int = 0

    FIELDS = {
        "input_tokens": F(2, "int"),
        "output_tokens": F(3, "int"),
    }

    def encode(self) -> bytes:
        from ..protocol.capv2 import _tag, _enc_varint, _WT_LEN

        out = bytearray()
        for m in self.messages:
            b = m.encode()
            out += _tag(1, _WT_LEN) + _enc_varint(len(b)) + b
        out += super().encode()
        return bytes(out)


@dataclass(eq=False)
class UpdateMemoryRequest(Message):
    memory_id: str = ""
    logical_payload: bytes = b""
    model_response: bytes = b""
    mode: int = 0

    FIELDS = {
        "memory_id": F(1, "str"),
        "logical_payload": F(2, "bytes"),
        "model_response": F(3, "bytes"),
        "mode": F(4, "int"),
    }


@dataclass(eq=False)
class Empty(Message):
    FIELDS = {}


@dataclass(eq=False)
class ListSnapshotsResponse(Message):
    snapshots: List[str] = dc_field(default_factory=list)

    FIELDS = {"snapshots": F(1, "rep_str")}


# -- service implementation ----------------------------------------------------

_MODES = {0: "raw", 1: "raw", 2: "chat", 3: "rag"}


def _auth_ok(context, api_keys) -> bool:
    if not api_keys:
        return True
    md = dict(context.invocation_metadata())
    return md.get("x-api-key", "") in api_keys


def make_grpc_server(node, api_keys: Optional[List[str]] = None,
                     max_workers: int = 8, auth_tenant: str = "") -> grpc.Server:
    from ..runtime.context_engine import ContextEngine
    from ..utils.ids import new_trace_id, new_id

    api_keys = set(api_keys or [])
    # auth-context tenant override (gateway.go:4149-4159): a request that
    # omits org_id lands in the authenticated tenant, not a global default
    auth_tenant = auth_tenant or os.environ.get("TENANT_ID", "")
    ctx_engine = ContextEngine(node.memory)

    def submit_job(req_bytes, context):
        if not _auth_ok(context, api_keys):
            context.abort(grpc.StatusCode.UNAUTHENTICATED, "invalid api key")
        req = SubmitJobRequest.decode(req_bytes)
        if not req.prompt:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "prompt is required")
        topic = req.topic or "job.default"
        if not topic.startswith("job."):
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "topic must start with job.")
        org = req.org_id or auth_tenant or "default"
        job_id = new_id()
        if req.idempotency_key:
            inserted, existing = node.job_store.try_set_idempotency_key(org, req.idempotency_key, job_id)
            if not inserted:
                meta = node.job_store.get_job_meta(existing)
                return SubmitJobResponse(job_id=existing, trace_id=meta.get("trace_id", "")).encode()
        jr = JobRequest(
            job_id=job_id,
            topic=topic,
            priority={"critical": JobPriority.CRITICAL, "interactive": JobPriority.INTERACTIVE,
                      "batch": JobPriority.BATCH}.get(req.priority.lower(), JobPriority.BATCH),
            adapter_id=req.adapter_id,
            memory_id=req.memory_id,
            tenant_id=org,
            principal_id=req.principal_id,
            labels=dict(req.labels),
            env={"tenant_id": org, "team_id": req.team_id},
            meta=JobMetadata(
                actor_id=req.actor_id,
                actor_type={"human": ActorType.HUMAN, "service": ActorType.SERVICE}.get(
                    req.actor_type.lower(), ActorType.UNSPECIFIED),
                idempotency_key=req.idempotency_key,
                capability=req.capability,
                risk_tags=list(req.risk_tags),
                requires=list(req.requires),
                pack_id=req.pack_id,
            ),
        )
        trace_id = new_trace_id()
        from ..protocol import JobState

        node.job_store.set_state(job_id, JobState.PENDING)
        node.job_store.set_job_meta(job_id, topic=topic, tenant=org, trace_id=trace_id)
        node.submit_job(jr, trace_id=trace_id,
                        context=json.dumps({"prompt": req.prompt}).encode())
        node.drain()
        return SubmitJobResponse(job_id=job_id, trace_id=trace_id).encode()

    def get_job_status(req_bytes, context):
        if not _auth_ok(context, api_keys):
            context.abort(grpc.StatusCode.UNAUTHENTICATED, "invalid api key")
        req = GetJobStatusRequest.decode(req_bytes)
        meta = node.job_store.get_job_meta(req.job_id)
        if not meta:
            context.abort(grpc.StatusCode.NOT_FOUND, "job not found")
        return GetJobStatusResponse(
            job_id=req.job_id,
            status=meta.get("state", ""),
            result_ptr=meta.get("result_ptr", ""),
        ).encode()

    def safety_eval(req_bytes, context):
        if not _auth_ok(context, api_keys):
            context.abort(grpc.StatusCode.UNAUTHENTICATED, "invalid api key")
        req = PolicyCheckRequest.decode(req_bytes)
        return node.safety_kernel.evaluate(req).encode()

    def list_snapshots(req_bytes, context):
        return ListSnapshotsResponse(snapshots=node.safety_kernel.list_snapshots()).encode()

    def build_window(req_bytes, context):
        req = BuildWindowRequest.decode(req_bytes)
        w = ctx_engine.build_window(
            req.memory_id, mode=_MODES.get(int(req.mode), "raw"),
            logical_payload=req.logical_payload,
            max_input_tokens=req.max_input_tokens or 8000,
            max_output_tokens=req.max_output_tokens or 1024,
        )
        return BuildWindowResponse(
            messages=[ModelMessagePb(role=m.role, content=m.content) for m in w.messages],
            input_tokens=w.input_tokens,
            output_tokens=w.output_tokens,
        ).encode()

    def update_memory(req_bytes, context):
        req = UpdateMemoryRequest.decode(req_bytes)
        ctx_engine.update_memory(req.memory_id, req.logical_payload, req.model_response,
                                 mode=_MODES.get(int(req.mode), "chat"))
        return Empty().encode()

    def unary(fn):
        return grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=lambda b: b, response_serializer=lambda b: b
        )

    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers((
        grpc.method_handlers_generic_handler("cordum.v1.CordumApi", {
            "SubmitJob": unary(submit_job),
            "GetJobStatus": unary(get_job_status),
        }),
        grpc.method_handlers_generic_handler("cordum.v1.SafetyKernel", {
            "Check": unary(safety_eval),
            "Evaluate": unary(safety_eval),
            "Explain": unary(safety_eval),
            "Simulate": unary(safety_eval),
            "ListSnapshots": unary(list_snapshots),
        }),
        grpc.method_handlers_generic_handler("cordum.v1.ContextEngine", {
            "BuildWindow": unary(build_window),
            "UpdateMemory": unary(update_memory),
        }),
    ))
    return server


def serve_grpc(node, address: str = "127.0.0.1:9090", api_keys: Optional[List[str]] = None,
               auth_tenant: str = "") -> grpc.Server:
    server = make_grpc_server(node, api_keys, auth_tenant=auth_tenant)
    server.add_insecure_port(address)
    server.start()
    return server
