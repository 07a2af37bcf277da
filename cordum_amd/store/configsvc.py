"""Overlay configuration service.

Oracle: core/configsvc/service.go:14-168 — scoped docs `cfg:<scope>:<id>`
(system/org/team/workflow/step), revision counters, `Effective()` shallow-merge
system -> org -> team -> workflow -> step, EffectiveSnapshot{version
"sys:N|org:M|...", hash = sha256(canonical JSON)}.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Tuple

from ..utils.canonical_json import canonical_json_hash

SCOPES = ("system", "org", "team", "workflow", "step")


@dataclass
class EffectiveSnapshot:
    config: Dict[str, Any]
    version: str
    hash: str


class ConfigService:
    def __init__(self):
        self._mu = threading.RLock()
        self._docs: Dict[Tuple[str, str], Dict[str, Any]] = {}
        self._revs: Dict[Tuple[str, str], int] = {}
        self._watchers: List = []

    def set(self, scope: str, doc_id: str, doc: Dict[str, Any]) -> int:
        if scope not in SCOPES:
            raise ValueError(f"bad scope {scope}")
        with self._mu:
            key = (scope, doc_id)
            self._docs[key] = dict(doc)
            self._revs[key] = self._revs.get(key, 0) + 1
            rev = self._revs[key]
            watchers = list(self._watchers)
        for w in watchers:
            try:
                w(scope, doc_id)
            except Exception:
                pass
        return rev

    def patch(self, scope: str, doc_id: str, patch: Dict[str, Any]) -> int:
        """RFC 7386 JSON merge patch into an existing doc (pack overlays)."""
        with self._mu:
            cur = dict(self._docs.get((scope, doc_id), {}))
        merged = json_merge_patch(cur, patch)
        return self.set(scope, doc_id, merged)

    def get(self, scope: str, doc_id: str) -> Optional[Dict[str, Any]]:
        with self._mu:
            doc = self._docs.get((scope, doc_id))
            return dict(doc) if doc is not None else None

    def revision(self, scope: str, doc_id: str) -> int:
        with self._mu:
            return self._revs.get((scope, doc_id), 0)

    def watch(self, callback) -> None:
        """callback(scope, doc_id) fires synchronously after each write."""
        with self._mu:
            self._watchers.append(callback)

    def effective(
        self,
        org: str = "",
        team: str = "",
        workflow: str = "",
        step: str = "",
        system_id: str = "default",
    ) -> EffectiveSnapshot:
        """Shallow merge per top-level category key (service.go:117-149)."""
        layers = [("system", system_id), ("org", org), ("team", team), ("workflow", workflow), ("step", step)]
        merged: Dict[str, Any] = {}
        version_parts = []
        with self._mu:
            for scope, doc_id in layers:
                if not doc_id:
                    continue
                doc = self._docs.get((scope, doc_id))
                rev = self._revs.get((scope, doc_id), 0)
                version_parts.append(f"{scope[:3] if scope != 'system' else 'sys'}:{rev}")
                if doc:
                    for k, v in doc.items():
                        if isinstance(v, dict) and isinstance(merged.get(k), dict):
                            nv = dict(merged[k])
                            nv.update(v)
                            merged[k] = nv
                        else:
                            merged[k] = v
        version = "|".join(version_parts)
        return EffectiveSnapshot(config=merged, version=version, hash=canonical_json_hash(merged))

    def snapshot(self) -> Dict[str, Any]:
        with self._mu:
            return {f"cfg:{s}:{i}": dict(d) for (s, i), d in self._docs.items()}

    def restore(self, snap: Dict[str, Any]) -> None:
        for key, doc in snap.items():
            if key.startswith("cfg:"):
                _, scope, doc_id = key.split(":", 2)
                self.set(scope, doc_id, doc)


def json_merge_patch(target: Any, patch: Any) -> Any:
    """RFC 7386."""
    if not isinstance(patch, dict):
        return patch
    if not isinstance(target, dict):
        target = {}
    out = dict(target)
    for k, v in patch.items():
        if v is None:
            out.pop(k, None)
        else:
            out[k] = json_merge_patch(out.get(k), v)
    return out
