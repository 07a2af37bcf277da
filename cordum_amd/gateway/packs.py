"""Pack system: install/uninstall/verify/list + marketplace endpoints.

Oracle: gateway/packs.go — pack = .tgz bundle with `pack.yaml` manifest
(apiVersion cordum.io/v1alpha1, metadata, compatibility, topics, resources
{schemas, workflows}, overlays{config: json_merge_patch into cfg docs;
policy: bundle_fragment into cfg:system:policy.bundles}, tests), size limits
(64 MiB upload, 2048 files, 32 MiB/file, 256 MiB uncompressed, zip-slip path
rejection — docs/pack.md:85-92), install with plan/digest-noop detection and
a rollback stack (:608-813), soft uninstall (disable routing/policy, keep
workflows/schemas — docs/pack.md:12), verify (:887), registry doc
`cfg:system:packs`, marketplace catalogs `cfg:system:pack_catalogs` with
sha256-pinned URL installs (:454-607).
"""
from __future__ import annotations

import hashlib
import io
import json
import tarfile
from typing import Any, Callable, Dict, List

import yaml
from fastapi import APIRouter, Depends, HTTPException, Request

API_VERSION = "cordum.io/v1alpha1"
MAX_UPLOAD = 64 << 20
MAX_FILES = 2048
MAX_FILE = 32 << 20
MAX_UNCOMPRESSED = 256 << 20

PACKS_DOC = ("system", "packs")
CATALOGS_DOC = ("system", "pack_catalogs")
POLICY_DOC = ("system", "policy")


class PackError(Exception):
    pass


def extract_pack(blob: bytes) -> Dict[str, bytes]:
    """Untar with the reference's limits; rejects path traversal."""
    if len(blob) > MAX_UPLOAD:
        raise PackError("pack too large")
    files: Dict[str, bytes] = {}
    total = 0
    try:
        with tarfile.open(fileobj=io.BytesIO(blob), mode="r:*") as tf:
            for member in tf:
                if not member.isfile():
                    continue
                # strip only a leading "./" prefix — a charwise lstrip("./")
                # would silently sanitize "../x" to "x" instead of rejecting
                name = member.name
                while name.startswith("./"):
                    name = name[2:]
                if name.startswith("/") or ".." in name.split("/") or not name:
                    raise PackError(f"illegal path {member.name!r}")
                if member.size > MAX_FILE:
                    raise PackError(f"file too large: {name}")
                total += member.size
                if total > MAX_UNCOMPRESSED:
                    raise PackError("pack uncompressed size limit exceeded")
                if len(files) >= MAX_FILES:
                    raise PackError("too many files in pack")
                f = tf.extractfile(member)
                files[name] = f.read() if f else b""
    except tarfile.TarError as e:
        raise PackError(f"invalid archive: {e}")
    return files


def load_manifest(files: Dict[str, bytes]) -> Dict[str, Any]:
    raw = files.get("pack.yaml") or files.get("pack.yml")
    if raw is None:
        raise PackError("pack.yaml missing")
    try:
        manifest = yaml.safe_load(raw)
    except yaml.YAMLError as e:
        raise PackError(f"invalid pack.yaml: {e}")
    if not isinstance(manifest, dict):
        raise PackError("pack.yaml must be a mapping")
    if manifest.get("apiVersion") != API_VERSION:
        raise PackError(f"unsupported apiVersion {manifest.get('apiVersion')!r}")
    meta = manifest.get("metadata") or {}
    # the reference manifest uses metadata.id (packs.go:140-146,
    # examples/hello-pack/pack.yaml); accept `name` as an alias
    pack_id = meta.get("id") or meta.get("name")
    if not pack_id:
        raise PackError("metadata.id required")
    meta["name"] = pack_id
    meta["id"] = pack_id
    compat = manifest.get("compatibility") or {}
    pv = compat.get("protocolVersion")
    if pv is not None and int(pv) != 1:
        raise PackError(f"incompatible protocolVersion {pv}")
    return manifest


def _resource_paths(entries) -> list:
    """resources.{schemas,workflows} entries are plain paths OR the
    reference's {id, path} objects (packs.go packResource)."""
    out = []
    for e in entries or []:
        if isinstance(e, str):
            out.append((None, e))
        elif isinstance(e, dict) and e.get("path"):
            out.append((e.get("id") or None, e["path"]))
        else:
            raise PackError(f"invalid resource entry: {e!r}")
    return out


def _digest(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


class PackInstaller:
    """Install/uninstall with a rollback stack (packs.go:703-813)."""

    def __init__(self, node):
        self.node = node

    def install(self, blob: bytes, source: str = "upload") -> Dict[str, Any]:
        files = extract_pack(blob)
        manifest = load_manifest(files)
        meta = manifest.get("metadata") or {}
        pack_id = meta["name"]

        locks = self.node.locks
        if not locks.acquire(f"pack:{pack_id}", "pack-installer", "exclusive", ttl_s=60):
            raise PackError("pack install already in progress")
        rollback: List[Callable[[], None]] = []
        plan = {"schemas": [], "workflows": [], "config_overlays": 0, "policy_overlays": 0}
        try:
            resources = manifest.get("resources") or {}
            # schemas
            for rid, path in _resource_paths(resources.get("schemas")):
                raw = files.get(path)
                if raw is None:
                    raise PackError(f"schema file missing: {path}")
                doc = _parse_doc(raw, path)
                sid = rid or doc.get("$id") or doc.get("id") or path.rsplit("/", 1)[-1].rsplit(".", 1)[0]
                prev = self.node.schemas.get(sid)
                if prev is not None and _digest(json.dumps(prev, sort_keys=True).encode()) == _digest(
                    json.dumps(doc, sort_keys=True).encode()
                ):
                    plan["schemas"].append({"id": sid, "action": "noop"})
                    continue
                self.node.schemas.put(sid, doc)
                plan["schemas"].append({"id": sid, "action": "update" if prev else "create"})
                rollback.append(lambda sid=sid, prev=prev: (
                    self.node.schemas.put(sid, prev) if prev is not None else self.node.schemas.delete(sid)
                ))
            # workflows
            from ..workflow import Workflow

            for _wf_rid, path in _resource_paths(resources.get("workflows")):
                raw = files.get(path)
                if raw is None:
                    raise PackError(f"workflow file missing: {path}")
                doc = _parse_doc(raw, path)
                wf = Workflow.from_dict(doc)
                if not wf.id:
                    raise PackError(f"workflow id missing in {path}")
                try:
                    prev = self.node.workflow_store.get_workflow(wf.id)
                except KeyError:
                    prev = None
                self.node.workflow_store.put_workflow(wf)
                plan["workflows"].append({"id": wf.id, "action": "update" if prev else "create"})
                rollback.append(lambda wf_id=wf.id, prev=prev: (
                    self.node.workflow_store.put_workflow(prev) if prev is not None
                    else self.node.workflow_store.delete_workflow(wf_id)
                ))
            # overlays
            overlays = manifest.get("overlays") or {}
            for ov in overlays.get("config") or []:
                scope = ov.get("scope", "system")
                key = ov.get("key", "default")
                patch = ov.get("json_merge_patch")
                if patch is None and ov.get("path"):
                    # reference overlay form: strategy + patch content in a
                    # bundled file (packs.go packConfigOverlay)
                    raw = files.get(ov["path"])
                    if raw is None:
                        raise PackError(f"overlay file missing: {ov['path']}")
                    patch = _parse_doc(raw, ov["path"])
                patch = patch or {}
                prev = self.node.configsvc.get(scope, key)
                self.node.configsvc.patch(scope, key, patch)
                plan["config_overlays"] += 1
                rollback.append(lambda scope=scope, key=key, prev=prev:
                                self.node.configsvc.set(scope, key, prev or {}))
            for i, ov in enumerate(overlays.get("policy") or []):
                frag = ov.get("bundle_fragment", "")
                if not frag and ov.get("path"):
                    raw = files.get(ov["path"])
                    if raw is None:
                        raise PackError(f"overlay file missing: {ov['path']}")
                    frag = raw.decode("utf-8")
                bundle_id = ov.get("id") or ov.get("name") or f"{pack_id}/{i}"
                if "/" not in bundle_id:
                    bundle_id = f"{pack_id}/{bundle_id}"
                prev_doc = self.node.configsvc.get(*POLICY_DOC) or {}
                self.node.configsvc.patch(*POLICY_DOC, {
                    "bundles": {bundle_id: {"enabled": True, "content": frag, "pack_id": pack_id}},
                })
                plan["policy_overlays"] += 1
                rollback.append(lambda prev_doc=prev_doc:
                                self.node.configsvc.set(*POLICY_DOC, prev_doc))
            # registry entry (cfg:system:packs)
            registry = self.node.configsvc.get(*PACKS_DOC) or {}
            packs = dict(registry.get("packs", {}))
            packs[pack_id] = {
                "id": pack_id,
                "version": str(meta.get("version", "")),
                "description": str(meta.get("description", "")),
                "status": "ACTIVE",
                "source": source,
                "digest": _digest(blob),
                "topics": manifest.get("topics") or [],
                "plan": plan,
            }
            registry["packs"] = packs
            self.node.configsvc.set(*PACKS_DOC, registry)
            return {"pack_id": pack_id, "status": "ACTIVE", "plan": plan}
        except Exception:
            for undo in reversed(rollback):
                try:
                    undo()
                except Exception:
                    pass
            raise
        finally:
            locks.release(f"pack:{pack_id}", "pack-installer")

    def uninstall(self, pack_id: str) -> Dict[str, Any]:
        """Soft uninstall: disable policy fragments + registry INACTIVE;
        workflows/schemas are kept (docs/pack.md:12)."""
        registry = self.node.configsvc.get(*PACKS_DOC) or {}
        packs = dict(registry.get("packs", {}))
        if pack_id not in packs:
            raise PackError(f"pack {pack_id!r} not installed")
        policy_doc = self.node.configsvc.get(*POLICY_DOC) or {}
        bundles = dict(policy_doc.get("bundles", {}))
        for bid, b in bundles.items():
            if isinstance(b, dict) and b.get("pack_id") == pack_id:
                nb = dict(b)
                nb["enabled"] = False
                bundles[bid] = nb
        policy_doc["bundles"] = bundles
        self.node.configsvc.set(*POLICY_DOC, policy_doc)
        entry = dict(packs[pack_id])
        entry["status"] = "INACTIVE"
        packs[pack_id] = entry
        registry["packs"] = packs
        self.node.configsvc.set(*PACKS_DOC, registry)
        return {"pack_id": pack_id, "status": "INACTIVE"}

    def verify(self, pack_id: str) -> Dict[str, Any]:
        registry = self.node.configsvc.get(*PACKS_DOC) or {}
        entry = (registry.get("packs") or {}).get(pack_id)
        if entry is None:
            raise PackError(f"pack {pack_id!r} not installed")
        problems = []
        for wf in (entry.get("plan") or {}).get("workflows", []):
            try:
                self.node.workflow_store.get_workflow(wf["id"])
            except KeyError:
                problems.append(f"workflow {wf['id']} missing")
        for sc in (entry.get("plan") or {}).get("schemas", []):
            if self.node.schemas.get(sc["id"]) is None:
                problems.append(f"schema {sc['id']} missing")
        return {"pack_id": pack_id, "ok": not problems, "problems": problems}


def _parse_doc(raw: bytes, path: str) -> Dict[str, Any]:
    try:
        if path.endswith((".yaml", ".yml")):
            doc = yaml.safe_load(raw)
        else:
            doc = json.loads(raw)
    except (yaml.YAMLError, ValueError) as e:
        raise PackError(f"invalid document {path}: {e}")
    if not isinstance(doc, dict):
        raise PackError(f"document {path} must be a mapping")
    return doc


def make_packs_router(node, principal_dep, admin_dep) -> APIRouter:
    router = APIRouter()
    installer = PackInstaller(node)

    @router.get("/packs")
    def list_packs(p=Depends(principal_dep)):
        registry = node.configsvc.get(*PACKS_DOC) or {}
        return {"items": list((registry.get("packs") or {}).values())}

    @router.get("/packs/{pack_id}")
    def get_pack(pack_id: str, p=Depends(principal_dep)):
        registry = node.configsvc.get(*PACKS_DOC) or {}
        entry = (registry.get("packs") or {}).get(pack_id)
        if entry is None:
            raise HTTPException(404, "pack not found")
        return entry

    @router.post("/packs/install")
    async def install_pack(request: Request, p=Depends(admin_dep)):
        blob = await request.body()
        if not blob:
            raise HTTPException(400, "pack archive required")
        try:
            return installer.install(blob)
        except PackError as e:
            raise HTTPException(400, str(e))

    @router.post("/packs/{pack_id}/uninstall")
    def uninstall_pack(pack_id: str, p=Depends(admin_dep)):
        try:
            return installer.uninstall(pack_id)
        except PackError as e:
            raise HTTPException(404, str(e))

    @router.post("/packs/{pack_id}/verify")
    def verify_pack(pack_id: str, p=Depends(principal_dep)):
        try:
            return installer.verify(pack_id)
        except PackError as e:
            raise HTTPException(404, str(e))

    # -- marketplace (packs.go:454-607, :933-1200) ---------------------------
    # Catalog registry doc cfg:system:pack_catalogs {"catalogs":[{id,title,
    # url,enabled}]}; each enabled catalog's file ({updated_at, packs:[...]})
    # is fetched through node.marketplace_fetcher(url) -> bytes (pluggable:
    # there is no egress in this deployment, so the default fetcher refuses;
    # tests and real deployments inject one — marketplace_test.go uses a stub
    # HTTP server the same way). Snapshot cached 30 s; URL installs must
    # match an enabled catalog entry AND its pinned sha256.
    mkt = Marketplace(node, installer)

    @router.get("/marketplace/packs")
    def marketplace(p=Depends(admin_dep)):
        return mkt.snapshot()

    @router.post("/marketplace/install")
    async def marketplace_install(request: Request, p=Depends(admin_dep)):
        body_raw = await request.body()
        try:
            body = json.loads(body_raw or b"{}")
        except ValueError:
            raise HTTPException(400, "invalid json")
        if not isinstance(body, dict):
            raise HTTPException(400, "invalid json payload")
        return mkt.install(body)

    return router


DEFAULT_CATALOG_URL = "https://packs.cordum.io/catalog.json"
MARKETPLACE_CACHE_TTL = 30.0


class Marketplace:
    """Catalog registry + 30 s snapshot cache + sha256-pinned installs
    (packs.go:454-607 handleMarketplaceInstall, :933 marketplaceSnapshot,
    :1128/:1168 findMarketplaceEntry/ByURL)."""

    def __init__(self, node, installer):
        self.node = node
        self.installer = installer
        self._cache = None
        self._cache_at = 0.0

    def _fetch(self, url: str) -> bytes:
        fetcher = getattr(self.node, "marketplace_fetcher", None)
        if fetcher is None:
            raise PackError("marketplace fetch unavailable (no egress; "
                            "configure a marketplace fetcher)")
        return fetcher(url)

    def _catalogs(self) -> List[Dict[str, Any]]:
        doc = self.node.configsvc.get(*CATALOGS_DOC) or {}
        cats = doc.get("catalogs")
        if cats is None:
            cats = [{"id": "cordum", "title": "Cordum Packs",
                     "url": DEFAULT_CATALOG_URL, "enabled": True}]
        return [c for c in cats if isinstance(c, dict)]

    def _load_entries(self):
        statuses, entries = [], []
        for idx, cat in enumerate(self._catalogs()):
            cid = (cat.get("id") or "").strip() or f"catalog-{idx + 1}"
            status = {
                "id": cid,
                "title": (cat.get("title") or "").strip(),
                "url": (cat.get("url") or "").strip(),
                "enabled": bool(cat.get("enabled", True)),
            }
            if not status["enabled"]:
                statuses.append(status)
                continue
            try:
                raw = self._fetch(status["url"])
                doc = json.loads(raw)
            except Exception as e:  # catalog errors are per-catalog, not fatal
                status["error"] = str(e)
                statuses.append(status)
                continue
            status["updated_at"] = doc.get("updated_at", "")
            statuses.append(status)
            for pack in doc.get("packs", []) or []:
                if isinstance(pack, dict):
                    entries.append({"pack": pack, "catalog_id": cid,
                                    "catalog_title": status["title"]})
        return statuses, entries

    def snapshot(self, refresh: bool = False) -> Dict[str, Any]:
        now = self.node.clock.now() if hasattr(self.node, "clock") else 0.0
        if not refresh and self._cache is not None and \
                now - self._cache_at < MARKETPLACE_CACHE_TTL:
            resp = dict(self._cache)
            resp["cached"] = True
            return resp
        statuses, entries = self._load_entries()
        items = []
        for e in entries:
            p = e["pack"]
            items.append({
                "id": p.get("id", ""), "version": p.get("version", ""),
                "title": p.get("title", ""), "description": p.get("description", ""),
                "author": p.get("author", ""), "homepage": p.get("homepage", ""),
                "source": p.get("source", ""), "license": p.get("license", ""),
                "url": p.get("url", ""), "sha256": p.get("sha256", ""),
                "capabilities": p.get("capabilities", []) or [],
                "requires": p.get("requires", []) or [],
                "risk_tags": p.get("risk_tags", []) or [],
                "catalog_id": e["catalog_id"],
                "catalog_title": e["catalog_title"],
            })
        resp = {"catalogs": statuses, "items": items, "fetched_at": now}
        self._cache, self._cache_at = resp, now
        return dict(resp)

    def _find_entry(self, catalog_id: str, pack_id: str, version: str):
        _, entries = self._load_entries()
        for e in entries:
            p = e["pack"]
            if e["catalog_id"] == catalog_id and p.get("id") == pack_id:
                if version and p.get("version") != version:
                    continue
                return e
        raise HTTPException(404, "marketplace entry not found")

    def _find_entry_by_url(self, url: str):
        _, entries = self._load_entries()
        for e in entries:
            if (e["pack"].get("url") or "").strip() == url:
                return e
        raise HTTPException(404, "pack url not found in any enabled catalog")

    def install(self, body: Dict[str, Any]) -> Dict[str, Any]:
        url = (body.get("url") or "").strip()
        sha = (body.get("sha256") or "").strip()
        if url:
            if not sha:
                raise HTTPException(400, "sha256 required")
            entry = self._find_entry_by_url(url)
            p = entry["pack"]
            entry_url = (p.get("url") or "").strip()
            entry_sha = (p.get("sha256") or "").strip()
            if not entry_url or not entry_sha:
                raise HTTPException(400, "marketplace entry missing url or sha256")
            if sha.lower() != entry_sha.lower():
                raise HTTPException(400, "sha256 mismatch")
            url, sha = entry_url, entry_sha
        else:
            catalog_id = (body.get("catalog_id") or "").strip()
            pack_id = (body.get("pack_id") or "").strip()
            if not catalog_id or not pack_id:
                raise HTTPException(400, "catalog_id and pack_id required")
            entry = self._find_entry(catalog_id, pack_id,
                                     (body.get("version") or "").strip())
            p = entry["pack"]
            url = (p.get("url") or "").strip()
            sha = (p.get("sha256") or "").strip()
        if not url:
            raise HTTPException(400, "download url required")
        if not sha:
            raise HTTPException(400, "sha256 required")
        try:
            blob = self._fetch(url)
        except PackError as e:
            raise HTTPException(400, str(e))
        except Exception as e:
            raise HTTPException(400, f"download failed: {e}")
        if _digest(blob).lower() != sha.lower():
            raise HTTPException(400, "sha256 mismatch")
        try:
            return self.installer.install(blob, source=f"marketplace:{url}")
        except PackError as e:
            raise HTTPException(400, str(e))
