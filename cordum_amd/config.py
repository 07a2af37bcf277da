"""Configuration loading: env registry + YAML file formats.

Oracle: core/infra/config/config.go:24-80 (env keys), pools.go:13-130
(pools.yaml: `topics: {topic: pool|[pools]}`, `pools: {name: {requires}}`),
timeouts.go:10-127 (timeouts.yaml `reconciler` section, defaults
300/9000/30 s), safety.yaml via safety/policy.parse_safety_policy, and
system.yaml seeded into the config service as cfg:system:default
(cmd/cordum-scheduler/config_overlay.go:28-110).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Dict, Optional

import yaml

from .scheduler.strategy import PoolRouting, routing_from_pools_yaml

DEFAULT_CONFIG_DIR = Path(__file__).resolve().parent.parent / "config"


@dataclass
class Timeouts:
    dispatch_timeout_s: float = 300.0
    running_timeout_s: float = 9000.0
    scan_interval_s: float = 30.0


@dataclass
class NodeConfig:
    """Env-derived configuration (infra/config/config.go Load())."""

    pool_config_path: str = ""
    timeout_config_path: str = ""
    safety_policy_path: str = ""
    system_config_path: str = ""
    api_rate_limit_rps: float = 0.0
    api_rate_limit_burst: int = 100
    safety_decision_cache_ttl_s: float = 30.0
    job_meta_ttl_s: float = 7 * 24 * 3600
    redis_data_ttl_s: float = 24 * 3600
    workflow_engine_scan_interval_s: float = 5.0
    scheduler_config_reload_interval_s: float = 30.0
    log_format_json: bool = False

    @classmethod
    def from_env(cls, config_dir: Optional[Path] = None) -> "NodeConfig":
        d = config_dir or DEFAULT_CONFIG_DIR

        def _f(name: str, default: float) -> float:
            try:
                return float(os.environ.get(name, default))
            except ValueError:
                return default

        return cls(
            pool_config_path=os.environ.get("POOL_CONFIG_PATH", str(d / "pools.yaml")),
            timeout_config_path=os.environ.get("TIMEOUT_CONFIG_PATH", str(d / "timeouts.yaml")),
            safety_policy_path=os.environ.get("SAFETY_POLICY_PATH", str(d / "safety.yaml")),
            system_config_path=os.environ.get("SYSTEM_CONFIG_PATH", str(d / "system.yaml")),
            api_rate_limit_rps=_f("API_RATE_LIMIT_RPS", 0.0),
            api_rate_limit_burst=int(_f("API_RATE_LIMIT_BURST", 100)),
            safety_decision_cache_ttl_s=_f("SAFETY_DECISION_CACHE_TTL", 30.0),
            job_meta_ttl_s=_f("JOB_META_TTL", 7 * 24 * 3600),
            redis_data_ttl_s=_f("REDIS_DATA_TTL", 24 * 3600),
            workflow_engine_scan_interval_s=_f("WORKFLOW_ENGINE_SCAN_INTERVAL", 5.0),
            scheduler_config_reload_interval_s=_f("SCHEDULER_CONFIG_RELOAD_INTERVAL", 30.0),
            log_format_json=os.environ.get("CORDUM_LOG_FORMAT", "").lower() == "json",
        )


def load_yaml(path: str) -> Dict[str, Any]:
    p = Path(path)
    if not path or not p.exists():
        return {}
    with open(p) as f:
        doc = yaml.safe_load(f)
    return doc if isinstance(doc, dict) else {}


def load_pools(path: str) -> PoolRouting:
    doc = load_yaml(path)
    if not doc:
        return PoolRouting(topics={"job.default": ["default"]}, pools={})
    return routing_from_pools_yaml(doc)


def load_timeouts(path: str) -> Timeouts:
    doc = load_yaml(path)
    rec = doc.get("reconciler") or {}
    return Timeouts(
        dispatch_timeout_s=float(rec.get("dispatch_timeout_seconds", 300) or 300),
        running_timeout_s=float(rec.get("running_timeout_seconds", 9000) or 9000),
        scan_interval_s=float(rec.get("scan_interval_seconds", 30) or 30),
    )


def load_safety_yaml(path: str, public_key: str = "", signature: str = "") -> str:
    """Load the safety policy file; verify its ed25519 signature when a key
    is configured (oracle: safetykernel bundle load + signature verify,
    kernel.go:787-868; env SAFETY_POLICY_PUBLIC_KEY / SAFETY_POLICY_SIGNATURE,
    hex or base64)."""
    p = Path(path)
    if not path or not p.exists():
        return ""
    data = p.read_bytes()
    public_key = public_key or os.environ.get("SAFETY_POLICY_PUBLIC_KEY", "")
    signature = signature or os.environ.get("SAFETY_POLICY_SIGNATURE", "")
    if public_key:
        from .utils.ed25519 import verify

        if not signature:
            sig_file = Path(str(p) + ".sig")
            if sig_file.exists():
                signature = sig_file.read_text().strip()
        if not signature:
            raise ValueError("safety policy signature required but missing")
        if not verify(_decode_key(public_key), _decode_key(signature), data):
            raise ValueError("safety policy signature verification failed")
    return data.decode("utf-8")


def _decode_key(s: str) -> bytes:
    import base64
    import binascii

    s = s.strip()
    try:
        return bytes.fromhex(s)
    except ValueError:
        pass
    try:
        return base64.b64decode(s, validate=True)
    except (binascii.Error, ValueError):
        return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def seed_system_config(configsvc, path: str) -> None:
    """Seed file config into cfg:system:default if absent
    (config_overlay.go:28-110)."""
    doc = load_yaml(path)
    if not doc:
        return
    if configsvc.get("system", "default") is None:
        configsvc.set("system", "default", doc)
