"""Deterministic job hashing for approval binding.

Oracle: core/controlplane/scheduler/job_hash.go:15-48 — clone the JobRequest,
strip `approval_*` labels, the bus msg-id label, and the effective-config env
var, marshal deterministically, SHA-256. Approvals are bound to
(policy snapshot, job hash); the scheduler re-checks on re-entry
(scheduler/engine.go:484-522) and the gateway verifies before approving
(gateway.go:3763-3784).
"""
from __future__ import annotations

import hashlib

from ..protocol import capv2

BUS_MSG_ID_LABEL = "cordum.bus_msg_id"
EFFECTIVE_CONFIG_ENV = "CORDUM_EFFECTIVE_CONFIG"


def sha256_hex(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


def job_hash(req: capv2.JobRequest) -> str:
    clone = req.copy()
    # job_hash.go:27 lowercases the key before the prefix check, so mixed-case
    # Approval_* labels are stripped too (cross-implementation hash parity)
    clone.labels = {
        k: v
        for k, v in clone.labels.items()
        if not k.lower().startswith("approval_") and k != BUS_MSG_ID_LABEL
    }
    clone.env = {k: v for k, v in clone.env.items() if k != EFFECTIVE_CONFIG_ENV}
    return sha256_hex(clone.encode())
