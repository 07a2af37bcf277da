"""CAP v2 message shapes + protobuf wire codec.

Message/field shapes mirror the reference control plane's use of the CAP v2
module (type aliases: core/protocol/pb/v1/pb.go:1-76; request build sites:
core/controlplane/scheduler/safety_client.go:68-106, core/workflow/engine.go:1320-1415,
core/infra/registry/snapshot.go). The upstream `github.com/cordum-io/cap/v2`
.proto files are not vendored in the reference repo, so the field NUMBERING
here is this project's own documented layout; the wire FORMAT is standard
protobuf (varint/length-delimited), encoded deterministically (ascending field
numbers, map entries sorted by key) so that SHA-256 job hashes
(utils/hashing.py, oracle core/controlplane/scheduler/job_hash.go:15-48) are
stable across processes and hosts.

Everything here is plain Python + bytes; the scheduler's hot path never
round-trips through this codec (device job descriptors are packed tensors —
see cordum_amd/ops/policy_compile.py JobEncoder and ops/pipeline.py). This
codec serves the API boundary, the WAL, and cross-process workers.
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field as dc_field, fields as dc_fields
from enum import IntEnum
from typing import Any, ClassVar, Dict, List, Optional, Tuple

# ---------------------------------------------------------------------------
# Enums (values match proto enum ordering used by the reference API surface)
# ---------------------------------------------------------------------------


class JobPriority(IntEnum):
    UNSPECIFIED = 0
    INTERACTIVE = 1
    BATCH = 2
    CRITICAL = 3


class JobStatus(IntEnum):
    """Proto job status (pb.go:52-62). Store-level states add APPROVAL_REQUIRED."""

    UNSPECIFIED = 0
    PENDING = 1
    SCHEDULED = 2
    DISPATCHED = 3
    RUNNING = 4
    SUCCEEDED = 5
    FAILED = 6
    CANCELLED = 7
    DENIED = 8
    TIMEOUT = 9


class ActorType(IntEnum):
    UNSPECIFIED = 0
    HUMAN = 1
    SERVICE = 2


class DecisionType(IntEnum):
    UNSPECIFIED = 0
    ALLOW = 1
    DENY = 2
    REQUIRE_HUMAN = 3
    THROTTLE = 4
    ALLOW_WITH_CONSTRAINTS = 5


# ---------------------------------------------------------------------------
# Minimal deterministic protobuf codec
# ---------------------------------------------------------------------------

_WT_VARINT = 0
_WT_I64 = 1
_WT_LEN = 2
_WT_I32 = 5


def _enc_varint(v: int) -> bytes:
    if v < 0:
        v &= (1 << 64) - 1
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _dec_varint(buf: bytes, i: int) -> Tuple[int, int]:
    shift = 0
    val = 0
    while True:
        b = buf[i]
        i += 1
        val |= (b & 0x7F) << shift
        if not b & 0x80:
            return val, i
        shift += 7
        if shift > 70:
            raise ValueError("varint overflow")


def _tag(num: int, wt: int) -> bytes:
    return _enc_varint((num << 3) | wt)


@dataclass(frozen=True)
class F:
    """Field spec: wire number, kind, and (for messages) the submessage type."""

    num: int
    kind: str  # int|sint64|bool|enum|str|bytes|double|msg|rep_str|map_ss
    sub: Any = None  # message class for kind == "msg"
    enum: Any = None  # IntEnum class for kind == "enum"


class Message:
    """Base for CAP messages: declarative FIELDS -> encode/decode/json."""

    FIELDS: ClassVar[Dict[str, F]] = {}

    # -- binary wire format --------------------------------------------------
    def encode(self) -> bytes:
        out = bytearray()
        for name, spec in sorted(self.FIELDS.items(), key=lambda kv: kv[1].num):
            val = getattr(self, name)
            out += _encode_field(spec, val)
        # unknown fields captured at decode re-emit verbatim (after known
        # fields, like Go protobuf): a node relaying packets from a newer
        # CAP client must not strip fields it doesn't model
        out += getattr(self, "_unknown", b"")
        return bytes(out)

    @classmethod
    def decode(cls, data: bytes) -> "Message":
        msg = cls()
        by_num = {spec.num: (name, spec) for name, spec in cls.FIELDS.items()}
        i, n = 0, len(data)
        while i < n:
            fstart = i
            key, i = _dec_varint(data, i)
            num, wt = key >> 3, key & 7
            if wt == _WT_VARINT:
                raw, i = _dec_varint(data, i)
            elif wt == _WT_LEN:
                ln, i = _dec_varint(data, i)
                raw = data[i : i + ln]
                i += ln
            elif wt == _WT_I64:
                raw = data[i : i + 8]
                i += 8
            elif wt == _WT_I32:
                raw = data[i : i + 4]
                i += 4
            else:
                raise ValueError(f"bad wire type {wt}")
            ent = by_num.get(num)
            if ent is None:
                # unknown field: preserve raw bytes for re-encode
                msg._unknown = getattr(msg, "_unknown", b"") + bytes(data[fstart:i])
                continue
            name, spec = ent
            if _WIRE_TYPE.get(spec.kind, _WT_LEN) != wt:
                raise ValueError(
                    f"wire type {wt} invalid for field {num} ({spec.kind})")
            _decode_into(msg, name, spec, raw)
        return msg

    # -- JSON (protojson-style camelCase; accepts snake_case too) ------------
    def to_dict(self, camel: bool = True) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        for name, spec in self.FIELDS.items():
            val = getattr(self, name)
            if _is_default(spec, val):
                continue
            key = _camel(name) if camel else name
            if spec.kind == "msg":
                d[key] = val.to_dict(camel)
            elif spec.kind == "enum":
                d[key] = _enum_json_name(spec.enum, val)
            else:
                d[key] = val
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Message":
        msg = cls()
        lookup = {}
        for name, spec in cls.FIELDS.items():
            lookup[name] = (name, spec)
            lookup[_camel(name)] = (name, spec)
        for key, val in (d or {}).items():
            ent = lookup.get(key)
            if ent is None:
                continue
            name, spec = ent
            if spec.kind == "msg":
                setattr(msg, name, spec.sub.from_dict(val or {}))
            elif spec.kind == "enum":
                setattr(msg, name, _enum_from_json(spec.enum, val))
            elif spec.kind in ("int", "sint64"):
                setattr(msg, name, int(val or 0))
            elif spec.kind == "double":
                setattr(msg, name, float(val or 0.0))
            elif spec.kind == "bool":
                setattr(msg, name, bool(val))
            elif spec.kind == "bytes":
                import base64

                setattr(msg, name, base64.b64decode(val) if isinstance(val, str) else bytes(val or b""))
            elif spec.kind == "rep_str":
                setattr(msg, name, list(val or []))
            elif spec.kind == "map_ss":
                setattr(msg, name, dict(val or {}))
            else:
                setattr(msg, name, str(val or ""))
        return msg

    def copy(self):
        return type(self).decode(self.encode())

    def __eq__(self, other):
        return type(self) is type(other) and self.encode() == other.encode()


_WIRE_TYPE = {
    "int": _WT_VARINT, "sint64": _WT_VARINT, "bool": _WT_VARINT,
    "enum": _WT_VARINT, "double": _WT_I64, "str": _WT_LEN, "bytes": _WT_LEN,
    "msg": _WT_LEN, "rep_str": _WT_LEN, "map_ss": _WT_LEN, "rep_msg": _WT_LEN,
}


def _is_default(spec: F, val: Any) -> bool:
    if spec.kind == "msg":
        return val is None
    if spec.kind in ("str",):
        return val == ""
    if spec.kind == "bytes":
        return not val
    if spec.kind in ("rep_str",):
        return not val
    if spec.kind == "map_ss":
        return not val
    if spec.kind == "double":
        return val == 0.0
    if spec.kind == "bool":
        return not val
    return int(val) == 0


def _encode_field(spec: F, val: Any) -> bytes:
    if _is_default(spec, val):
        return b""
    k = spec.kind
    if k in ("int", "enum", "bool", "sint64"):
        return _tag(spec.num, _WT_VARINT) + _enc_varint(int(val))
    if k == "double":
        return _tag(spec.num, _WT_I64) + struct.pack("<d", float(val))
    if k == "str":
        b = val.encode("utf-8")
        return _tag(spec.num, _WT_LEN) + _enc_varint(len(b)) + b
    if k == "bytes":
        return _tag(spec.num, _WT_LEN) + _enc_varint(len(val)) + bytes(val)
    if k == "msg":
        b = val.encode()
        return _tag(spec.num, _WT_LEN) + _enc_varint(len(b)) + b
    if k == "rep_str":
        out = bytearray()
        for s in val:
            b = s.encode("utf-8")
            out += _tag(spec.num, _WT_LEN) + _enc_varint(len(b)) + b
        return bytes(out)
    if k == "map_ss":
        out = bytearray()
        for mk in sorted(val):  # deterministic
            kv = bytearray()
            kb = mk.encode("utf-8")
            vb = val[mk].encode("utf-8")
            # map entries serialize BOTH fields even when empty (protobuf
            # MapEntry semantics — verified byte-for-byte against the
            # google.protobuf runtime in tests/test_capv2_interop.py)
            kv += _tag(1, _WT_LEN) + _enc_varint(len(kb)) + kb
            kv += _tag(2, _WT_LEN) + _enc_varint(len(vb)) + vb
            out += _tag(spec.num, _WT_LEN) + _enc_varint(len(kv)) + bytes(kv)
        return bytes(out)
    raise ValueError(f"unknown kind {k}")


def _decode_into(msg: "Message", name: str, spec: F, raw: Any) -> None:
    k = spec.kind
    if k in ("int", "sint64"):
        v = raw if raw < (1 << 63) else raw - (1 << 64)
        setattr(msg, name, v)
    elif k == "bool":
        setattr(msg, name, bool(raw))
    elif k == "enum":
        try:
            setattr(msg, name, spec.enum(raw))
        except ValueError:
            setattr(msg, name, raw)
    elif k == "double":
        setattr(msg, name, struct.unpack("<d", raw)[0])
    elif k == "str":
        setattr(msg, name, raw.decode("utf-8"))
    elif k == "bytes":
        setattr(msg, name, bytes(raw))
    elif k == "msg":
        setattr(msg, name, spec.sub.decode(raw))
    elif k == "rep_str":
        getattr(msg, name).append(raw.decode("utf-8"))
    elif k == "map_ss":
        key_s, val_s = "", ""
        i, n = 0, len(raw)
        while i < n:
            tag, i = _dec_varint(raw, i)
            ln, i = _dec_varint(raw, i)
            s = raw[i : i + ln].decode("utf-8")
            i += ln
            if tag >> 3 == 1:
                key_s = s
            else:
                val_s = s
        getattr(msg, name)[key_s] = val_s


def _camel(s: str) -> str:
    parts = s.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


def _enum_json_name(enum_cls, v) -> Any:
    try:
        member = enum_cls(v)
    except ValueError:
        return int(v)
    prefix = {
        "JobPriority": "JOB_PRIORITY_",
        "JobStatus": "JOB_STATUS_",
        "ActorType": "ACTOR_TYPE_",
        "DecisionType": "DECISION_TYPE_",
    }.get(enum_cls.__name__, "")
    return prefix + member.name


def _enum_from_json(enum_cls, v) -> Any:
    if isinstance(v, int):
        try:
            return enum_cls(v)
        except ValueError:
            return v
    s = str(v)
    for member in enum_cls:
        if s == member.name or s.endswith("_" + member.name) or s.upper() == member.name:
            return member
    return enum_cls(0)


# ---------------------------------------------------------------------------
# Messages
# ---------------------------------------------------------------------------


@dataclass(eq=False)
class Budget(Message):
    max_tokens: int = 0
    deadline_ms: int = 0

    FIELDS = {"max_tokens": F(1, "int"), "deadline_ms": F(2, "int")}


@dataclass(eq=False)
class ContextHints(Message):
    max_input_tokens: int = 0
    max_output_tokens: int = 0
    context_mode: str = ""

    FIELDS = {
        "max_input_tokens": F(1, "int"),
        "max_output_tokens": F(2, "int"),
        "context_mode": F(3, "str"),
    }


@dataclass(eq=False)
class JobMetadata(Message):
    actor_id: str = ""
    actor_type: ActorType = ActorType.UNSPECIFIED
    idempotency_key: str = ""
    capability: str = ""
    risk_tags: List[str] = dc_field(default_factory=list)
    requires: List[str] = dc_field(default_factory=list)
    pack_id: str = ""
    labels: Dict[str, str] = dc_field(default_factory=dict)

    FIELDS = {
        "actor_id": F(1, "str"),
        "actor_type": F(2, "enum", enum=ActorType),
        "idempotency_key": F(3, "str"),
        "capability": F(4, "str"),
        "risk_tags": F(5, "rep_str"),
        "requires": F(6, "rep_str"),
        "pack_id": F(7, "str"),
        "labels": F(8, "map_ss"),
    }


@dataclass(eq=False)
class JobRequest(Message):
    job_id: str = ""
    topic: str = ""
    priority: JobPriority = JobPriority.UNSPECIFIED
    context_ptr: str = ""
    adapter_id: str = ""
    memory_id: str = ""
    tenant_id: str = ""
    principal_id: str = ""
    labels: Dict[str, str] = dc_field(default_factory=dict)
    env: Dict[str, str] = dc_field(default_factory=dict)
    meta: Optional[JobMetadata] = None
    context_hints: Optional[ContextHints] = None
    budget: Optional[Budget] = None
    workflow_id: str = ""
    parent_job_id: str = ""

    FIELDS = {
        "job_id": F(1, "str"),
        "topic": F(2, "str"),
        "priority": F(3, "enum", enum=JobPriority),
        "context_ptr": F(4, "str"),
        "adapter_id": F(5, "str"),
        "memory_id": F(6, "str"),
        "tenant_id": F(7, "str"),
        "principal_id": F(8, "str"),
        "labels": F(9, "map_ss"),
        "env": F(10, "map_ss"),
        "meta": F(11, "msg", sub=JobMetadata),
        "context_hints": F(12, "msg", sub=ContextHints),
        "budget": F(13, "msg", sub=Budget),
        "workflow_id": F(14, "str"),
        "parent_job_id": F(15, "str"),
    }


@dataclass(eq=False)
class JobResult(Message):
    job_id: str = ""
    status: JobStatus = JobStatus.UNSPECIFIED
    result_ptr: str = ""
    worker_id: str = ""
    execution_ms: int = 0
    error_code: str = ""
    error_message: str = ""

    FIELDS = {
        "job_id": F(1, "str"),
        "status": F(2, "enum", enum=JobStatus),
        "result_ptr": F(3, "str"),
        "worker_id": F(4, "str"),
        "execution_ms": F(5, "int"),
        "error_code": F(6, "str"),
        "error_message": F(7, "str"),
    }


@dataclass(eq=False)
class JobProgress(Message):
    job_id: str = ""
    worker_id: str = ""
    progress: float = 0.0
    message: str = ""

    FIELDS = {
        "job_id": F(1, "str"),
        "worker_id": F(2, "str"),
        "progress": F(3, "double"),
        "message": F(4, "str"),
    }


@dataclass(eq=False)
class JobCancel(Message):
    job_id: str = ""
    reason: str = ""

    FIELDS = {"job_id": F(1, "str"), "reason": F(2, "str")}


@dataclass(eq=False)
class Heartbeat(Message):
    worker_id: str = ""
    region: str = ""
    type: str = ""
    cpu_load: float = 0.0
    gpu_utilization: float = 0.0
    active_jobs: int = 0
    capabilities: List[str] = dc_field(default_factory=list)
    pool: str = ""
    max_parallel_jobs: int = 0
    labels: Dict[str, str] = dc_field(default_factory=dict)

    FIELDS = {
        "worker_id": F(1, "str"),
        "region": F(2, "str"),
        "type": F(3, "str"),
        "cpu_load": F(4, "double"),
        "gpu_utilization": F(5, "double"),
        "active_jobs": F(6, "int"),
        "capabilities": F(7, "rep_str"),
        "pool": F(8, "str"),
        "max_parallel_jobs": F(9, "int"),
        "labels": F(10, "map_ss"),
    }


@dataclass(eq=False)
class SystemAlert(Message):
    severity: str = ""
    message: str = ""
    source: str = ""
    labels: Dict[str, str] = dc_field(default_factory=dict)

    FIELDS = {
        "severity": F(1, "str"),
        "message": F(2, "str"),
        "source": F(3, "str"),
        "labels": F(4, "map_ss"),
    }


# -- policy ------------------------------------------------------------------


@dataclass(eq=False)
class BudgetConstraints(Message):
    max_runtime_ms: int = 0
    max_retries: int = 0
    max_artifact_bytes: int = 0
    max_concurrent_jobs: int = 0

    FIELDS = {
        "max_runtime_ms": F(1, "int"),
        "max_retries": F(2, "int"),
        "max_artifact_bytes": F(3, "int"),
        "max_concurrent_jobs": F(4, "int"),
    }


@dataclass(eq=False)
class SandboxProfile(Message):
    isolated: bool = False
    network_allowlist: List[str] = dc_field(default_factory=list)
    filesystem: str = ""

    FIELDS = {
        "isolated": F(1, "bool"),
        "network_allowlist": F(2, "rep_str"),
        "filesystem": F(3, "str"),
    }


@dataclass(eq=False)
class ToolchainConstraints(Message):
    allowed_tools: List[str] = dc_field(default_factory=list)
    allowed_commands: List[str] = dc_field(default_factory=list)

    FIELDS = {"allowed_tools": F(1, "rep_str"), "allowed_commands": F(2, "rep_str")}


@dataclass(eq=False)
class DiffConstraints(Message):
    max_files: int = 0
    max_lines: int = 0
    deny_paths: List[str] = dc_field(default_factory=list)

    FIELDS = {
        "max_files": F(1, "int"),
        "max_lines": F(2, "int"),
        "deny_paths": F(3, "rep_str"),
    }


@dataclass(eq=False)
class PolicyConstraints(Message):
    budgets: Optional[BudgetConstraints] = None
    sandbox: Optional[SandboxProfile] = None
    toolchain: Optional[ToolchainConstraints] = None
    diff: Optional[DiffConstraints] = None
    redaction_level: str = ""

    FIELDS = {
        "budgets": F(1, "msg", sub=BudgetConstraints),
        "sandbox": F(2, "msg", sub=SandboxProfile),
        "toolchain": F(3, "msg", sub=ToolchainConstraints),
        "diff": F(4, "msg", sub=DiffConstraints),
        "redaction_level": F(5, "str"),
    }


@dataclass(eq=False)
class PolicyRemediation(Message):
    id: str = ""
    description: str = ""
    replacement_topic: str = ""
    replacement_capability: str = ""
    add_labels: Dict[str, str] = dc_field(default_factory=dict)
    remove_labels: List[str] = dc_field(default_factory=list)

    FIELDS = {
        "id": F(1, "str"),
        "description": F(2, "str"),
        "replacement_topic": F(3, "str"),
        "replacement_capability": F(4, "str"),
        "add_labels": F(5, "map_ss"),
        "remove_labels": F(6, "rep_str"),
    }


@dataclass(eq=False)
class PolicyCheckRequest(Message):
    job_id: str = ""
    topic: str = ""
    tenant: str = ""
    principal_id: str = ""
    priority: JobPriority = JobPriority.UNSPECIFIED
    budget: Optional[Budget] = None
    labels: Dict[str, str] = dc_field(default_factory=dict)
    memory_id: str = ""
    meta: Optional[JobMetadata] = None
    effective_config: bytes = b""

    FIELDS = {
        "job_id": F(1, "str"),
        "topic": F(2, "str"),
        "tenant": F(3, "str"),
        "principal_id": F(4, "str"),
        "priority": F(5, "enum", enum=JobPriority),
        "budget": F(6, "msg", sub=Budget),
        "labels": F(7, "map_ss"),
        "memory_id": F(8, "str"),
        "meta": F(9, "msg", sub=JobMetadata),
        "effective_config": F(10, "bytes"),
    }


@dataclass(eq=False)
class PolicyCheckResponse(Message):
    decision: DecisionType = DecisionType.UNSPECIFIED
    reason: str = ""
    rule_id: str = ""
    policy_snapshot: str = ""
    constraints: Optional[PolicyConstraints] = None
    approval_required: bool = False
    approval_ref: str = ""
    remediations: List[PolicyRemediation] = dc_field(default_factory=list)

    FIELDS = {
        "decision": F(1, "enum", enum=DecisionType),
        "reason": F(2, "str"),
        "rule_id": F(3, "str"),
        "policy_snapshot": F(4, "str"),
        "constraints": F(5, "msg", sub=PolicyConstraints),
        "approval_required": F(6, "bool"),
        "approval_ref": F(7, "str"),
        "remediations": F(8, "rep_msg"),
    }

    # repeated message needs custom handling
    def encode(self) -> bytes:
        out = bytearray()
        for name, spec in sorted(self.FIELDS.items(), key=lambda kv: kv[1].num):
            val = getattr(self, name)
            if spec.kind == "rep_msg":
                for m in val:
                    b = m.encode()
                    out += _tag(spec.num, _WT_LEN) + _enc_varint(len(b)) + b
            else:
                out += _encode_field(spec, val)
        out += getattr(self, "_unknown", b"")
        return bytes(out)

    @classmethod
    def decode(cls, data: bytes) -> "PolicyCheckResponse":
        msg = cls()
        i, n = 0, len(data)
        by_num = {spec.num: (name, spec) for name, spec in cls.FIELDS.items()}
        while i < n:
            fstart = i
            key, i = _dec_varint(data, i)
            num, wt = key >> 3, key & 7
            if wt == _WT_VARINT:
                raw, i = _dec_varint(data, i)
            elif wt == _WT_LEN:
                ln, i = _dec_varint(data, i)
                raw = data[i : i + ln]
                i += ln
            elif wt == _WT_I64:
                raw = data[i : i + 8]
                i += 8
            elif wt == _WT_I32:
                raw = data[i : i + 4]
                i += 4
            else:
                raise ValueError(f"bad wire type {wt}")
            ent = by_num.get(num)
            if ent is None:
                msg._unknown = getattr(msg, "_unknown", b"") + bytes(data[fstart:i])
                continue
            name, spec = ent
            if _WIRE_TYPE.get(spec.kind, _WT_LEN) != wt:
                raise ValueError(
                    f"wire type {wt} invalid for field {num} ({spec.kind})")
            if spec.kind == "rep_msg":
                msg.remediations.append(PolicyRemediation.decode(raw))
            else:
                _decode_into(msg, name, spec, raw)
        return msg

    def to_dict(self, camel: bool = True) -> Dict[str, Any]:
        d = {}
        for name, spec in self.FIELDS.items():
            val = getattr(self, name)
            key = _camel(name) if camel else name
            if spec.kind == "rep_msg":
                if val:
                    d[key] = [m.to_dict(camel) for m in val]
            elif not _is_default(spec, val):
                if spec.kind == "msg":
                    d[key] = val.to_dict(camel)
                elif spec.kind == "enum":
                    d[key] = _enum_json_name(spec.enum, val)
                else:
                    d[key] = val
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "PolicyCheckResponse":
        msg = super().from_dict({k: v for k, v in (d or {}).items() if k not in ("remediations",)})
        for r in (d or {}).get("remediations", []) or []:
            msg.remediations.append(PolicyRemediation.from_dict(r))
        return msg


@dataclass(eq=False)
class BusPacket(Message):
    """The bus envelope (oneof payload encoded as distinct fields; at most one set)."""

    trace_id: str = ""
    tenant_id: str = ""
    # proto3 implicit presence: the wire default MUST be 0 (a 1 default
    # would re-materialize on decode(encode(x)) of a foreign packet and
    # break byte round-trips — caught by tests/test_capv2_interop.py).
    # Senders set protocol_version=1 explicitly (capsdk constants).
    protocol_version: int = 0
    labels: Dict[str, str] = dc_field(default_factory=dict)
    job_request: Optional[JobRequest] = None
    job_result: Optional[JobResult] = None
    heartbeat: Optional[Heartbeat] = None
    alert: Optional[SystemAlert] = None
    job_progress: Optional[JobProgress] = None
    job_cancel: Optional[JobCancel] = None

    FIELDS = {
        "trace_id": F(1, "str"),
        "tenant_id": F(2, "str"),
        "protocol_version": F(3, "int"),
        "labels": F(4, "map_ss"),
        "job_request": F(10, "msg", sub=JobRequest),
        "job_result": F(11, "msg", sub=JobResult),
        "heartbeat": F(12, "msg", sub=Heartbeat),
        "alert": F(13, "msg", sub=SystemAlert),
        "job_progress": F(14, "msg", sub=JobProgress),
        "job_cancel": F(15, "msg", sub=JobCancel),
    }

    def payload(self):
        for name in ("job_request", "job_result", "heartbeat", "alert", "job_progress", "job_cancel"):
            v = getattr(self, name)
            if v is not None:
                return name, v
        return None, None

    def job_id(self) -> str:
        name, p = self.payload()
        return getattr(p, "job_id", "") if p is not None else ""


# Store-level state string <-> proto status mapping
# (workflowengine/reconciler.go:160-175 jobStatusFromState).
_STATE_TO_STATUS = {
    "PENDING": JobStatus.PENDING,
    "SCHEDULED": JobStatus.SCHEDULED,
    "DISPATCHED": JobStatus.DISPATCHED,
    "RUNNING": JobStatus.RUNNING,
    "SUCCEEDED": JobStatus.SUCCEEDED,
    "FAILED": JobStatus.FAILED,
    "CANCELLED": JobStatus.CANCELLED,
    "DENIED": JobStatus.DENIED,
    "TIMEOUT": JobStatus.TIMEOUT,
    "APPROVAL_REQUIRED": JobStatus.PENDING,
}


def status_from_state(state_name: str) -> JobStatus:
    return _STATE_TO_STATUS.get(state_name, JobStatus.UNSPECIFIED)


def state_name_from_status(status: JobStatus) -> str:
    if status == JobStatus.UNSPECIFIED:
        return ""
    return JobStatus(status).name


# ---------------------------------------------------------------------------
# Native codec engine (cordum_amd/protocol/native/capv2_codec.cpp)
# ---------------------------------------------------------------------------
# The C++ engine is schema-driven from the SAME FIELDS tables, so the wire
# layout has one source of truth; tests/test_property.py asserts byte
# equality against the pure-Python path on random messages. Loading is
# best-effort: without the built .so every path below stays pure Python.

_NATIVE = None
_KIND_CODES = {"int": 0, "sint64": 1, "bool": 2, "enum": 3, "str": 4,
               "bytes": 5, "double": 6, "msg": 7, "rep_str": 8, "map_ss": 9}


def _all_message_classes():
    seen, out, stack = set(), [], [Message]
    while stack:
        cls = stack.pop()
        for sub in cls.__subclasses__():
            if sub not in seen:
                seen.add(sub)
                out.append(sub)
                stack.append(sub)
    return out


def _load_native():
    global _NATIVE
    if _NATIVE is not None:
        return _NATIVE
    try:
        import importlib.util
        from pathlib import Path

        so = Path(__file__).resolve().parent / "native" / "_capv2_native.so"
        if not so.exists():
            return None
        spec = importlib.util.spec_from_file_location("_capv2_native", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        for cls in _all_message_classes():
            if any(fs.kind not in _KIND_CODES for fs in cls.FIELDS.values()):
                continue  # custom-kind classes (rep_msg) keep their own codec
            fields = [
                (name, fs.num, _KIND_CODES[fs.kind],
                 fs.sub.__name__ if fs.sub is not None else "",
                 fs.enum)
                for name, fs in cls.FIELDS.items()
            ]
            mod.register_message(cls.__name__, cls, fields)
        _NATIVE = mod
        return mod
    except Exception:
        return None


def _native_encode(msg: "Message"):
    n = _NATIVE
    if n is None:
        return None
    try:
        return n.encode(type(msg).__name__, msg)
    except Exception:
        return None


def _native_decode(cls, data: bytes):
    n = _NATIVE
    if n is None:
        return None
    try:
        return n.decode(cls.__name__, bytes(data))
    except Exception:
        return None


_PY_ENCODE = Message.encode
_PY_DECODE = Message.decode.__func__


def _encode_dispatch(self) -> bytes:
    out = _native_encode(self)
    if out is not None:
        return out
    return _PY_ENCODE(self)


def _decode_dispatch(cls, data: bytes):
    out = _native_decode(cls, data)
    if out is not None:
        return out
    return _PY_DECODE(cls, data)


Message.encode = _encode_dispatch
Message.decode = classmethod(_decode_dispatch)
Message.encode_py = _PY_ENCODE            # pure-Python paths kept addressable
Message.decode_py = classmethod(_PY_DECODE)  # (equality tests + fallback)

_load_native()
