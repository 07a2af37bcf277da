"""Durability: checkpoint/restore of all node state, Redis-compatible layout.

The reference's durability is "every mutation synchronously in Redis" +
JetStream file streams (SURVEY.md §5 checkpoint/resume). Here HBM/host state
is volatile, so durability is a host-side checkpoint whose KEY NAMING matches
the reference's Redis keys (`job:meta:<id>`, `wf:run:<id>`, `cfg:<scope>:<id>`,
`dlq:entry:<id>`, ...) so existing tooling can read it, plus an append-only
WAL of submitted job requests for replay between checkpoints.

Crash-safe restart = load the latest checkpoint, replay the WAL tail through
the scheduler (handlers are idempotent: terminal-state dedup + msg-id dedup,
exactly the at-least-once contract the reference relies on).
"""
from __future__ import annotations

import base64
import json
import os
import tempfile
import threading
from pathlib import Path

from ..protocol.capv2 import JobRequest


class Checkpointer:
    def __init__(self, node, directory: str):
        self.node = node
        self.dir = Path(directory)
        self.dir.mkdir(parents=True, exist_ok=True)
        self._wal_path = self.dir / "wal.jsonl"
        self._wal_file = None
        # Serializes wal_append against checkpoint: without it a submission
        # WAL-appended after the snapshot was built would be deleted by the
        # truncate and lost on crash-restore (gateway handlers run on a
        # threadpool, so appends are concurrent with checkpoints).
        self._lock = threading.Lock()

    # -- checkpoint -----------------------------------------------------------
    def checkpoint(self) -> str:
        """Write a full snapshot atomically; truncates the WAL.

        Holds the WAL lock across quiesce+snapshot+truncate so no append can
        land between the snapshot build and the truncate."""
        with self._lock:
            return self._checkpoint_locked()

    def _checkpoint_locked(self) -> str:
        # quiesce: every WAL-appended submission is pumped into the stores
        # before the snapshot is built, so truncating the WAL loses nothing
        drain = getattr(self.node, "drain", None)
        if callable(drain):
            drain()
        snap = {
            "version": 1,
            "jobs": self.node.job_store.snapshot(),
            "workflows": self.node.workflow_store.snapshot(),
            "config": self.node.configsvc.snapshot(),
            "dlq": self.node.dlq.snapshot(),
            "memory": {
                k: [base64.b64encode(b).decode(), e]
                for k, (b, e) in self.node.memory.snapshot().items()
            },
        }
        path = self.dir / "checkpoint.json"
        fd, tmp = tempfile.mkstemp(dir=str(self.dir), prefix=".ckpt-")
        try:
            with os.fdopen(fd, "w") as f:
                json.dump(snap, f)
                f.flush()
                os.fsync(f.fileno())
            os.replace(tmp, path)
        finally:
            if os.path.exists(tmp):
                os.unlink(tmp)
        self._truncate_wal()
        return str(path)

    def restore(self) -> bool:
        path = self.dir / "checkpoint.json"
        if not path.exists():
            return False
        with open(path) as f:
            snap = json.load(f)
        self.node.job_store.restore(snap.get("jobs", {}))
        self.node.workflow_store.restore(snap.get("workflows", {}))
        self.node.configsvc.restore(snap.get("config", {}))
        self.node.dlq.restore(snap.get("dlq", {}))
        mem = {
            k: (base64.b64decode(b), float(e))
            for k, (b, e) in snap.get("memory", {}).items()
        }
        self.node.memory.restore(mem)
        return True

    # -- WAL -------------------------------------------------------------------
    def wal_append(self, req: JobRequest, trace_id: str = "") -> None:
        with self._lock:
            if self._wal_file is None:
                self._wal_file = open(self._wal_path, "a")
            rec = {"trace_id": trace_id, "req": base64.b64encode(req.encode()).decode()}
            self._wal_file.write(json.dumps(rec) + "\n")
            self._wal_file.flush()
            os.fsync(self._wal_file.fileno())

    def _truncate_wal(self) -> None:
        if self._wal_file is not None:
            self._wal_file.close()
            self._wal_file = None
        if self._wal_path.exists():
            self._wal_path.unlink()

    def replay_wal(self) -> int:
        """Re-drive WAL-recorded submissions through the scheduler."""
        if not self._wal_path.exists():
            return 0
        n = 0
        with open(self._wal_path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    rec = json.loads(line)
                    req = JobRequest.decode(base64.b64decode(rec["req"]))
                except (ValueError, KeyError):
                    continue
                try:
                    self.node.scheduler.handle_job_request(req, rec.get("trace_id", ""))
                    n += 1
                except Exception:
                    pass
        self.node.drain()
        return n
