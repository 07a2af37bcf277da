"""Policy-bundle studio: CRUD, simulate-on-draft, snapshots, publish/rollback.

Oracle: gateway/policy_bundles.go — bundles live in the `cfg:system:policy`
doc's `bundles` map (id may contain '/' -> '~' path escape :24-122; admin
overlays under `secops/`), per-bundle enable toggle + author/message
metadata, simulate against draft content (:322-370), snapshots in
`cfg:system:policy_snapshots` (capture/list/get), publish/rollback with an
audit log in `cfg:system:policy_audit` (:370-525). The safety kernel's
config watch picks changes up immediately (in-process; the reference polls
every 30 s, kernel.go:485-508).
"""
from __future__ import annotations

import json
from typing import Any, Dict

from fastapi import APIRouter, Depends, HTTPException, Request

from ..safety import parse_safety_policy, policy as pol
from ..utils.ids import new_id

POLICY_DOC = ("system", "policy")
SNAPSHOTS_DOC = ("system", "policy_snapshots")
AUDIT_DOC = ("system", "policy_audit")


def unescape_bundle_id(path_id: str) -> str:
    return path_id.replace("~", "/")


def escape_bundle_id(bundle_id: str) -> str:
    return bundle_id.replace("/", "~")


async def _json_body(request: Request) -> Dict[str, Any]:
    raw = await request.body()
    try:
        body = json.loads(raw or b"{}")
    except ValueError:
        raise HTTPException(400, "invalid json")
    if not isinstance(body, dict):
        raise HTTPException(400, "object body required")
    return body


def make_policy_bundles_router(node, principal_dep, admin_dep) -> APIRouter:
    router = APIRouter()

    def _doc() -> Dict[str, Any]:
        return node.configsvc.get(*POLICY_DOC) or {"enabled": True, "bundles": {}}

    def _audit(action: str, who: str, detail: Dict[str, Any]):
        doc = node.configsvc.get(*AUDIT_DOC) or {"entries": []}
        entries = list(doc.get("entries", []))
        entries.append({
            "id": new_id(), "action": action, "by": who,
            "at": node.clock.now_micros(), **detail,
        })
        node.configsvc.set(*AUDIT_DOC, {"entries": entries[-500:]})

    @router.get("/policy/bundles")
    def list_bundles(p=Depends(principal_dep)):
        doc = _doc()
        items = []
        for bid, b in (doc.get("bundles") or {}).items():
            items.append({
                "id": bid, "enabled": b.get("enabled", True),
                "author": b.get("author", ""), "message": b.get("message", ""),
                "pack_id": b.get("pack_id", ""),
                "size": len(b.get("content", "") or ""),
            })
        return {"items": items, "enabled": doc.get("enabled", True)}

    @router.get("/policy/bundles/{bundle_id:path}")
    def get_bundle(bundle_id: str, p=Depends(principal_dep)):
        bid = unescape_bundle_id(bundle_id)
        b = (_doc().get("bundles") or {}).get(bid)
        if b is None:
            raise HTTPException(404, "bundle not found")
        return {"id": bid, **b}

    @router.put("/policy/bundles/{bundle_id:path}")
    async def put_bundle(bundle_id: str, request: Request, p=Depends(admin_dep)):
        bid = unescape_bundle_id(bundle_id)
        body = await _json_body(request)
        content = body.get("content", "")
        if content:
            try:
                parsed = parse_safety_policy(content)
            except Exception as e:
                raise HTTPException(400, f"invalid bundle: {e}")
            if parsed is None:
                raise HTTPException(400, "empty bundle content")
        doc = _doc()
        bundles = dict(doc.get("bundles") or {})
        bundles[bid] = {
            "enabled": bool(body.get("enabled", True)),
            "content": content,
            "author": body.get("author", getattr(p, "id", "")),
            "message": body.get("message", ""),
        }
        doc["bundles"] = bundles
        node.configsvc.set(*POLICY_DOC, doc)
        _audit("bundle_put", getattr(p, "id", ""), {"bundle_id": bid})
        return {"id": bid, "snapshot": node.safety_kernel.snapshot}

    @router.post("/policy/bundles/{bundle_id:path}/simulate")
    async def simulate_bundle(bundle_id: str, request: Request, p=Depends(principal_dep)):
        """Draft simulate: throwaway policy from draft + other enabled bundles
        (policy_bundles.go:322-370), no side effects."""
        bid = unescape_bundle_id(bundle_id)
        body = await _json_body(request)
        draft = body.get("content", "")
        inputs = body.get("inputs") or [body.get("input") or {}]
        merged = pol.SafetyPolicy()
        doc = _doc()
        for other_id in sorted(doc.get("bundles") or {}):
            b = doc["bundles"][other_id]
            content = draft if other_id == bid else b.get("content", "")
            if other_id != bid and not b.get("enabled", True):
                continue
            try:
                frag = parse_safety_policy(content)
            except Exception:
                continue
            if frag is None:
                continue
            merged.rules.extend(frag.rules)
            merged.tenants.update(frag.tenants)
        if bid not in (doc.get("bundles") or {}):
            try:
                frag = parse_safety_policy(draft)
                if frag is not None:
                    merged.rules.extend(frag.rules)
                    merged.tenants.update(frag.tenants)
            except Exception as e:
                raise HTTPException(400, f"invalid draft: {e}")
        results = []
        for inp_body in inputs:
            inp = pol.PolicyInput(
                tenant=inp_body.get("tenant", "default"),
                topic=inp_body.get("topic", ""),
                labels=dict(inp_body.get("labels") or {}),
                capability=inp_body.get("capability", ""),
                risk_tags=list(inp_body.get("risk_tags") or []),
                requires=list(inp_body.get("requires") or []),
                secrets_present=bool(inp_body.get("secrets_present", False)),
            )
            d = merged.evaluate(inp)
            results.append({
                "decision": d.decision, "rule_id": d.rule_id, "reason": d.reason,
                "approval_required": d.approval_required,
            })
        return {"results": results}

    @router.get("/policy/bundles/snapshots")
    def list_bundle_snapshots(p=Depends(principal_dep)):
        doc = node.configsvc.get(*SNAPSHOTS_DOC) or {"snapshots": {}}
        items = [
            {"id": sid, "at": s.get("at"), "by": s.get("by"), "message": s.get("message", "")}
            for sid, s in (doc.get("snapshots") or {}).items()
        ]
        items.sort(key=lambda s: s.get("at") or 0, reverse=True)
        return {"items": items}

    @router.post("/policy/bundles/snapshots")
    async def capture_snapshot(request: Request, p=Depends(admin_dep)):
        body = await _json_body(request)
        sid = new_id()
        doc = node.configsvc.get(*SNAPSHOTS_DOC) or {"snapshots": {}}
        snaps = dict(doc.get("snapshots") or {})
        snaps[sid] = {
            "at": node.clock.now_micros(),
            "by": getattr(p, "id", ""),
            "message": body.get("message", ""),
            "policy_doc": _doc(),
        }
        node.configsvc.set(*SNAPSHOTS_DOC, {"snapshots": snaps})
        _audit("snapshot_captured", getattr(p, "id", ""), {"snapshot_id": sid})
        return {"id": sid}

    @router.get("/policy/bundles/snapshots/{snap_id}")
    def get_snapshot(snap_id: str, p=Depends(principal_dep)):
        doc = node.configsvc.get(*SNAPSHOTS_DOC) or {"snapshots": {}}
        s = (doc.get("snapshots") or {}).get(snap_id)
        if s is None:
            raise HTTPException(404, "snapshot not found")
        return {"id": snap_id, **s}

    @router.post("/policy/publish")
    async def publish(request: Request, p=Depends(admin_dep)):
        """Capture a snapshot of the current bundles, then mark published."""
        body = await _json_body(request)
        sid = new_id()
        doc = node.configsvc.get(*SNAPSHOTS_DOC) or {"snapshots": {}}
        snaps = dict(doc.get("snapshots") or {})
        snaps[sid] = {
            "at": node.clock.now_micros(), "by": getattr(p, "id", ""),
            "message": body.get("message", "publish"), "policy_doc": _doc(),
            "published": True,
        }
        node.configsvc.set(*SNAPSHOTS_DOC, {"snapshots": snaps})
        _audit("published", getattr(p, "id", ""), {"snapshot_id": sid})
        return {"snapshot_id": sid, "kernel_snapshot": node.safety_kernel.snapshot}

    @router.post("/policy/rollback")
    async def rollback(request: Request, p=Depends(admin_dep)):
        body = await _json_body(request)
        sid = (body.get("snapshot_id") or "").strip()
        doc = node.configsvc.get(*SNAPSHOTS_DOC) or {"snapshots": {}}
        s = (doc.get("snapshots") or {}).get(sid)
        if s is None:
            raise HTTPException(404, "snapshot not found")
        node.configsvc.set(*POLICY_DOC, s.get("policy_doc") or {})
        _audit("rollback", getattr(p, "id", ""), {"snapshot_id": sid})
        return {"snapshot_id": sid, "kernel_snapshot": node.safety_kernel.snapshot}

    @router.get("/policy/audit")
    def audit_log(p=Depends(principal_dep)):
        doc = node.configsvc.get(*AUDIT_DOC) or {"entries": []}
        return {"items": list(reversed(doc.get("entries", [])))}

    return router
