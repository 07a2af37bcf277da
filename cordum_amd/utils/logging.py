"""Leveled key=value logger with optional JSON output.

Oracle: core/infra/logging/logging.go:18-100 — `level component msg k=v ...`
lines, `CORDUM_LOG_FORMAT=json` switches to JSON objects; components thread
trace_id/run_id/job_id fields.
"""
from __future__ import annotations

import json
import os
import sys
import threading
import time
from typing import Any, TextIO

LEVELS = {"debug": 10, "info": 20, "warn": 30, "error": 40}

_mu = threading.Lock()
_min_level = LEVELS.get(os.environ.get("CORDUM_LOG_LEVEL", "info").lower(), 20)
_json_mode = os.environ.get("CORDUM_LOG_FORMAT", "").lower() == "json"
_out: TextIO = sys.stderr


def configure(level: str = "info", json_format: bool = False, out: TextIO = None) -> None:
    global _min_level, _json_mode, _out
    _min_level = LEVELS.get(level.lower(), 20)
    _json_mode = json_format
    if out is not None:
        _out = out


def _log(level: str, component: str, msg: str, **fields: Any) -> None:
    if LEVELS[level] < _min_level:
        return
    ts = time.strftime("%Y-%m-%dT%H:%M:%S")
    with _mu:
        if _json_mode:
            rec = {"ts": ts, "level": level, "component": component, "msg": msg, **fields}
            _out.write(json.dumps(rec, default=str) + "\n")
        else:
            kv = " ".join(f"{k}={v}" for k, v in fields.items())
            _out.write(f"{ts} {level.upper():5s} {component}: {msg}{' ' + kv if kv else ''}\n")
        _out.flush()


def debug(component: str, msg: str, **fields):
    _log("debug", component, msg, **fields)


def info(component: str, msg: str, **fields):
    _log("info", component, msg, **fields)


def warn(component: str, msg: str, **fields):
    _log("warn", component, msg, **fields)


def error(component: str, msg: str, **fields):
    _log("error", component, msg, **fields)
