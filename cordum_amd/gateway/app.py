"""API gateway: the compat HTTP surface over the single-process node.

Route table oracle: gateway.go:701-805; response shapes:
job detail gateway.go:1011-1173, list envelopes {items, next_cursor} with
micros cursors :918-1009, submit validation/defaults :232-292 (prompt
required, topic must start `job.`, ≤50 labels/tags, defaults 8000/1024
tokens, topic job.default), approvals :3644-3925 (snapshot+hash-bound,
admin), remediation :1594-1755, DLQ retry :3452-3553, status :852-916,
WS stream :2002-2049 (protojson packets).

Implementation: FastAPI over runtime/node.Node. The reference's Go
mux+handlers become route functions; everything stays in-process (no Redis,
no NATS) — handlers call the node's stores/engines directly and drain the
in-process bus at request boundaries so synchronous clients observe
completed work (the reference gets the same effect from its synchronous
Redis writes).
"""
from __future__ import annotations

import json
import time
from typing import Any, Dict, List, Optional

from fastapi import (APIRouter, Depends, FastAPI, HTTPException, Query,
                     Request, Response, WebSocket, WebSocketDisconnect)
from fastapi.responses import JSONResponse

from ..protocol import JobState
from ..protocol import subjects as subj
from ..protocol.capv2 import (
    ActorType,
    JobStatus,
    Budget,
    BusPacket,
    ContextHints,
    JobMetadata,
    JobPriority,
    JobRequest,
    PolicyCheckRequest,
)
from ..runtime.node import Node
from ..store.memory_store import pointer_for_key
from ..utils.hashing import BUS_MSG_ID_LABEL, job_hash
from ..utils.ids import new_id, new_trace_id, short_id
from ..utils.secrets import contains_secret_refs
from ..workflow import Workflow, WorkflowRun
from ..store.job_store import ApprovalRecord
from .auth import BasicAuthProvider, Principal, TokenBucket

MAX_BODY_BYTES = 2 << 20  # 2 MiB (gateway.go:55)
MAX_PROMPT_CHARS = 100_000  # gateway.go:56

START_TIME = time.time()


def parse_actor_type(raw: str) -> ActorType:
    return {"human": ActorType.HUMAN, "service": ActorType.SERVICE}.get(
        (raw or "").strip().lower(), ActorType.UNSPECIFIED
    )


def create_app(
    node: Node,
    auth: Optional[BasicAuthProvider] = None,
    rate_limit_rps: float = 0.0,
    rate_limit_burst: int = 100,
    dashboard_dir: str = "",
    extensions: Optional[List[Any]] = None,
) -> FastAPI:
    app = FastAPI(title="cordum-mi355x gateway", version="0.1.0")
    # CORS for the dashboard (gateway.go:2051-2143)
    from fastapi.middleware.cors import CORSMiddleware

    app.add_middleware(
        CORSMiddleware, allow_origins=["*"], allow_methods=["*"],
        allow_headers=["*"], expose_headers=["*"],
    )
    auth = auth or BasicAuthProvider()
    bucket = TokenBucket(rate_limit_rps, rate_limit_burst, clock=node.clock)
    app.state.node = node
    app.state.auth = auth
    ws_clients: List[Any] = []

    # stream taps (gateway.go:531-649): job + audit events -> WS broadcast
    def _ws_tap(subject: str, pkt: BusPacket):
        if not ws_clients:
            return
        payload = {"subject": subject, "packet": _packet_json(pkt)}
        dead = []
        for q in ws_clients:
            try:
                q.append(payload)
            except Exception:
                dead.append(q)
        for q in dead:
            ws_clients.remove(q)

    node.bus.subscribe("sys.job.>", _ws_tap)
    node.bus.subscribe("sys.audit.>", _ws_tap)
    node.bus.subscribe(subj.SUBJECT_WORKFLOW_EVENT, _ws_tap)

    # enterprise extension seams (gateway/extensions.go:9-38): probe each
    # extension object for the protocols it implements
    from .extensions import (AuditEvent, AuditExporter, LicenseInfoProvider,
                             PublicPathProvider, RouteRegistrar,
                             path_is_public)

    exts = list(extensions or [])
    public_paths: List[str] = []
    license_provider: Optional[Any] = None
    for e in exts:
        if isinstance(e, PublicPathProvider):
            public_paths.extend(e.public_paths())
        if isinstance(e, LicenseInfoProvider):
            license_provider = e
    app.state.public_paths = public_paths
    app.state.license_provider = license_provider
    audit_exporters = [e for e in exts if isinstance(e, AuditExporter)]
    if audit_exporters:
        def _audit_tap(subject: str, pkt: BusPacket):
            alert = pkt.alert
            ev = AuditEvent(
                subject=subject,
                actor=getattr(alert, "source", "") if alert else "",
                action=getattr(alert, "severity", "") if alert else "",
                resource=pkt.trace_id,
                detail=_packet_json(pkt),
            )
            for ex in audit_exporters:
                try:
                    ex.export_audit(ev)
                except Exception:
                    pass  # a broken exporter must not break the bus

        node.bus.subscribe("sys.audit.>", _audit_tap)

    def principal(request: Request) -> Principal:
        if not bucket.allow():
            raise HTTPException(429, "rate limited")
        if public_paths and path_is_public(request.url.path, public_paths):
            return Principal(id="public", tenant="", role="public")
        p = auth.authenticate(dict(request.headers))
        if p is None:
            raise HTTPException(401, "invalid api key")
        return p

    def admin(request: Request) -> Principal:
        p = principal(request)
        if not auth.require_role(p, "admin"):
            raise HTTPException(403, "admin role required")
        return p

    api = APIRouter(prefix="/api/v1")

    # ------------------------------------------------------------------ jobs
    @api.post("/jobs")
    async def submit_job(request: Request, p: Principal = Depends(principal)):
        raw = await request.body()
        if len(raw) > MAX_BODY_BYTES:
            raise HTTPException(413, "body too large")
        try:
            body = json.loads(raw or b"{}")
        except ValueError:
            raise HTTPException(400, "invalid json")
        if not isinstance(body, dict):
            raise HTTPException(400, "body must be a json object")
        out = _submit_one(body, p)
        node.drain()
        return out

    @api.post("/jobs/batch")
    async def submit_jobs_batch(request: Request, p: Principal = Depends(principal)):
        """Batched ingest (MI355X-native addition): one HTTP request admits
        up to 4096 jobs; per-job semantics are identical to POST /jobs (same
        validation, idempotency, secrets scan, persistence) and the whole
        batch is flushed through the dispatch engine once — this is the
        HTTP-side amortization that feeds the batched device tick."""
        raw = await request.body()
        if len(raw) > 16 * MAX_BODY_BYTES:
            raise HTTPException(413, "body too large")
        try:
            body = json.loads(raw or b"{}")
        except ValueError:
            raise HTTPException(400, "invalid json")
        jobs = body.get("jobs") if isinstance(body, dict) else None
        if not isinstance(jobs, list) or not jobs:
            raise HTTPException(400, "jobs array required")
        if len(jobs) > 4096:
            raise HTTPException(400, "too many jobs (max 4096)")
        items = []
        for jb in jobs:
            if not isinstance(jb, dict):
                raise HTTPException(400, "every job must be a json object")
            try:
                items.append(_submit_one(jb, p))
            except HTTPException as e:
                items.append({"error": e.detail, "status": e.status_code})
        node.drain()
        return {"items": items}

    def _submit_one(body: Dict[str, Any], p: Principal) -> Dict[str, Any]:
        prompt = body.get("prompt") or ""
        if not prompt:
            raise HTTPException(400, "prompt is required")
        if len(prompt) > MAX_PROMPT_CHARS:
            raise HTTPException(400, f"prompt too long (>{MAX_PROMPT_CHARS} chars)")
        topic = body.get("topic") or "job.default"
        if not topic.startswith("job."):
            raise HTTPException(400, "topic must start with job.")
        labels = dict(body.get("labels") or {})
        tags = list(body.get("tags") or [])
        if len(labels) > 50:
            raise HTTPException(400, "too many labels (max 50)")
        if len(tags) > 50:
            raise HTTPException(400, "too many tags (max 50)")
        for k in ("max_input_tokens", "max_output_tokens", "max_total_tokens", "deadline_ms"):
            if int(body.get(k) or 0) < 0:
                raise HTTPException(400, f"{k} must be non-negative")
        actor_type_raw = body.get("actor_type") or ""
        if actor_type_raw and parse_actor_type(actor_type_raw) == ActorType.UNSPECIFIED:
            raise HTTPException(400, "actor_type must be 'human' or 'service'")

        org = body.get("org_id") or body.get("tenant_id") or auth.resolve_tenant(p, "")
        max_input = int(body.get("max_input_tokens") or 0) or 8000
        max_output = int(body.get("max_output_tokens") or 0) or 1024

        job_id = new_id()
        # scoped idempotency (gateway.go:1787-1812)
        idem = (body.get("idempotency_key") or "").strip()
        if idem:
            inserted, existing = node.job_store.try_set_idempotency_key(org, idem, job_id)
            if not inserted:
                meta = node.job_store.get_job_meta(existing)
                return {"job_id": existing, "trace_id": meta.get("trace_id", ""), "deduplicated": True}

        risk_tags = list(body.get("risk_tags") or [])
        context_payload = {"prompt": prompt}
        if tags:
            context_payload["tags"] = tags
        if body.get("context") is not None:
            context_payload["context"] = body["context"]
        # secrets scan (gateway.go:1820-1827)
        if contains_secret_refs(context_payload) or contains_secret_refs(labels):
            if "secrets" not in risk_tags:
                risk_tags.append("secrets")
            labels["secrets_present"] = "true"

        req = JobRequest(
            job_id=job_id,
            topic=topic,
            priority={"critical": JobPriority.CRITICAL, "interactive": JobPriority.INTERACTIVE,
                      "batch": JobPriority.BATCH}.get((body.get("priority") or "").lower(),
                                                      JobPriority.BATCH),
            adapter_id=body.get("adapter_id") or "",
            memory_id=body.get("memory_id") or "",
            tenant_id=org,
            principal_id=body.get("principal_id") or p.id,
            labels=labels,
            env={"tenant_id": org, "team_id": body.get("team_id") or ""},
            meta=JobMetadata(
                actor_id=body.get("actor_id") or "",
                actor_type=parse_actor_type(actor_type_raw),
                idempotency_key=idem,
                capability=body.get("capability") or "",
                risk_tags=risk_tags,
                requires=list(body.get("requires") or []),
                pack_id=body.get("pack_id") or "",
            ),
            context_hints=ContextHints(max_input_tokens=max_input, max_output_tokens=max_output),
        )
        deadline_ms = int(body.get("deadline_ms") or 0)
        max_total = int(body.get("max_total_tokens") or 0)
        if deadline_ms or max_total:
            req.budget = Budget(max_tokens=max_total, deadline_ms=deadline_ms)

        trace_id = new_trace_id()
        node.job_store.set_state(job_id, JobState.PENDING)
        node.job_store.set_job_meta(
            job_id, topic=topic, tenant=org, trace_id=trace_id,
            team=req.env.get("team_id", ""),
            actor_id=req.meta.actor_id, actor_type=actor_type_raw,
            idempotency_key=idem, capability=req.meta.capability,
            pack_id=req.meta.pack_id, risk_tags=risk_tags,
            requires=list(req.meta.requires), principal=req.principal_id,
        )
        node.submit_job(req, trace_id=trace_id, context=json.dumps(context_payload).encode())
        return {"job_id": job_id, "trace_id": trace_id}

    @api.get("/jobs")
    def list_jobs(
        p: Principal = Depends(principal),
        state: str = "",
        topic: str = "",
        tenant: str = "",
        team: str = "",
        trace_id: str = "",
        updated_after: int = 0,
        updated_before: int = 0,
        limit: int = Query(50, le=500),
        cursor: Optional[int] = None,
    ):
        """gateway.go:918-1009 — filters state/topic/tenant/team/trace +
        updated-window, micros cursor. A state filter pages the per-state
        index directly (complete at any scale); trace_id lists the trace
        set; otherwise the recent index pages by score."""
        updated_after = _normalize_micros(updated_after)
        updated_before = _normalize_micros(updated_before)
        cursor = _normalize_micros(cursor) if cursor else None
        state_u = state.upper() if state else ""
        if trace_id:
            ids, next_cursor = node.job_store.get_trace(trace_id), None
        elif state_u:
            try:
                st = JobState[state_u]
            except KeyError:
                return {"items": [], "next_cursor": None}
            ids, next_cursor = node.job_store.list_jobs_by_state_page(
                st, cursor=cursor, limit=limit)
        else:
            ids, next_cursor = node.job_store.list_recent(limit=limit, cursor=cursor)
        items = []
        for jid in ids:
            meta = node.job_store.get_job_meta(jid)
            if state_u and meta.get("state", "") != state_u:
                continue
            if topic and meta.get("topic", "") != topic:
                continue
            if tenant and meta.get("tenant", "") != tenant:
                continue
            if team and meta.get("team", "") != team:
                continue
            up = meta.get("updated_at", 0)
            if updated_after and up < updated_after:
                continue
            if updated_before and up > updated_before:
                continue
            items.append(_job_summary(jid, meta))
            if len(items) >= limit:
                break
        return {"items": items, "next_cursor": next_cursor}

    @api.get("/jobs/{job_id}")
    def get_job(job_id: str, p: Principal = Depends(principal)):
        state = node.job_store.get_state(job_id)
        if state == JobState.UNSPECIFIED:
            raise HTTPException(404, "job not found")
        return _job_detail(node, job_id)

    @api.get("/jobs/{job_id}/decisions")
    def job_decisions(job_id: str, p: Principal = Depends(principal)):
        rec = node.job_store.get_safety_decision(job_id)
        if rec is None:
            return {"items": []}
        return {"items": [_safety_record_json(rec)]}

    @api.post("/jobs/{job_id}/cancel")
    def cancel_job(job_id: str, p: Principal = Depends(principal)):
        ok = node.scheduler.cancel_job(job_id)
        node.drain()
        if not ok:
            raise HTTPException(409, "job not cancellable")
        return {"job_id": job_id, "state": str(node.job_store.get_state(job_id))}

    @api.post("/jobs/{job_id}/remediate")
    async def remediate_job(job_id: str, request: Request, p: Principal = Depends(principal)):
        """gateway.go:1594-1755."""
        body = await _json_body(request)
        rec = node.job_store.get_safety_decision(job_id)
        if rec is None or not rec.remediations:
            raise HTTPException(404, "no remediations available")
        rem_id = (body.get("remediation_id") or "").strip()
        if rem_id:
            rem = next((r for r in rec.remediations if r.id == rem_id), None)
        elif len(rec.remediations) == 1:
            rem = rec.remediations[0]
        else:
            raise HTTPException(400, "remediation_id required (multiple remediations)")
        if rem is None:
            raise HTTPException(404, "remediation not found")
        orig = node.job_store.get_job_request(job_id)
        if orig is None:
            raise HTTPException(404, "original job request not found")
        clone = JobRequest.decode(orig.encode())
        new_job_id = new_id()
        clone.job_id = new_job_id
        clone.parent_job_id = job_id
        if rem.replacement_topic:
            clone.topic = rem.replacement_topic
        if rem.replacement_capability:
            if clone.meta is None:
                clone.meta = JobMetadata()
            clone.meta.capability = rem.replacement_capability
        labels = {
            k: v for k, v in clone.labels.items()
            if not k.lower().startswith("approval_") and k != BUS_MSG_ID_LABEL
        }
        for k, v in rem.add_labels.items():
            labels[k] = v
        for k in rem.remove_labels:
            labels.pop(k, None)
        labels["remediation_of"] = job_id
        if rem.id:
            labels["remediation_id"] = rem.id
        labels[BUS_MSG_ID_LABEL] = f"remediation:{new_id()}"
        clone.labels = labels
        # copy context blob
        ctx = node.memory.get(f"ctx:{job_id}")
        node.job_store.set_state(new_job_id, JobState.PENDING)
        node.job_store.set_job_meta(new_job_id, topic=clone.topic, tenant=clone.tenant_id)
        node.submit_job(clone, context=ctx)
        node.drain()
        return {"job_id": new_job_id, "parent_job_id": job_id, "remediation_id": rem.id}

    # --------------------------------------------------------------- workers
    @api.get("/workers")
    def list_workers(p: Principal = Depends(principal)):
        return node.registry.cluster_snapshot()

    @api.get("/status")
    def status(p: Principal = Depends(principal)):
        return {
            "status": "ok",
            "uptime_sec": int(time.time() - START_TIME),
            "version": "0.1.0",
            "bus": "in-process",
            "store": "hbm+host",
            "workers": node.registry.count(),
            "policy_snapshot": node.safety_kernel.snapshot,
            "license": (license_provider.license_info()
                        if license_provider else {"edition": "oss"}),
        }

    # ---------------------------------------------------------------- memory
    @api.get("/memory")
    def read_memory(ptr: str, p: Principal = Depends(principal)):
        try:
            blob = node.memory.get_pointer(ptr)
        except ValueError:
            raise HTTPException(400, "invalid pointer")
        if blob is None:
            raise HTTPException(404, "not found")
        try:
            return JSONResponse(json.loads(blob))
        except ValueError:
            return Response(blob, media_type="application/octet-stream")

    # -------------------------------------------------------------- artifacts
    @api.post("/artifacts")
    async def put_artifact(request: Request, p: Principal = Depends(principal)):
        content = await request.body()
        retention = request.headers.get("x-retention", "standard")
        ptr = node.artifacts.put(
            content,
            content_type=request.headers.get("content-type", "application/octet-stream"),
            retention=retention,
        )
        return {"ptr": ptr}

    @api.get("/artifacts/{art_id}")
    def get_artifact(art_id: str, p: Principal = Depends(principal)):
        got = node.artifacts.get(art_id)
        if got is None:
            raise HTTPException(404, "artifact not found")
        blob, meta = got
        return Response(blob, media_type=meta.content_type)

    # ---------------------------------------------------------------- traces
    @api.get("/traces/{trace_id}")
    def get_trace(trace_id: str, p: Principal = Depends(principal)):
        jobs = node.job_store.get_trace(trace_id)
        return {
            "trace_id": trace_id,
            "jobs": [
                _job_summary(j, node.job_store.get_job_meta(j)) for j in jobs
            ],
        }

    # -------------------------------------------------------------- workflows
    @api.post("/workflows")
    async def create_workflow(request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        wf_id = (body.get("id") or "").strip() or new_id()
        body["id"] = wf_id
        if not body.get("steps"):
            raise HTTPException(400, "steps required")
        # field-preserving upsert (gateway.go:2534-2622); no cycle detection
        # (cyclic depends_on simply never becomes ready — documented behavior)
        existing = None
        try:
            existing = node.workflow_store.get_workflow(wf_id)
        except KeyError:
            pass
        wf = Workflow.from_dict(body)
        if existing is not None:
            wf.created_at = existing.created_at
            wf.created_by = wf.created_by or existing.created_by
        wf.org_id = wf.org_id or auth.resolve_tenant(p, "")
        node.workflow_store.put_workflow(wf)
        return wf.to_dict()

    @api.get("/workflows")
    def list_workflows(p: Principal = Depends(principal), org_id: str = ""):
        return {"items": [w.to_dict() for w in node.workflow_store.list_workflows(org_id)]}

    @api.get("/workflows/{wf_id}")
    def get_workflow(wf_id: str, p: Principal = Depends(principal)):
        try:
            return node.workflow_store.get_workflow(wf_id).to_dict()
        except KeyError:
            raise HTTPException(404, "workflow not found")

    @api.delete("/workflows/{wf_id}")
    def delete_workflow(wf_id: str, p: Principal = Depends(principal)):
        if not node.workflow_store.delete_workflow(wf_id):
            raise HTTPException(404, "workflow not found")
        return {"deleted": wf_id}

    @api.post("/workflows/{wf_id}/runs")
    async def start_run(wf_id: str, request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        try:
            wf = node.workflow_store.get_workflow(wf_id)
        except KeyError:
            raise HTTPException(404, "workflow not found")
        run_input = body.get("input") or {}
        # input-schema validation (gateway.go:2734-2739)
        if wf.input_schema:
            from ..store.schema_registry import validate_value

            errs = validate_value(wf.input_schema, run_input)
            if errs:
                raise HTTPException(400, f"input invalid: {'; '.join(errs)}")
        # run idempotency (gateway.go:2754-2782)
        idem = (body.get("idempotency_key") or request.headers.get("x-idempotency-key") or "").strip()
        run_id = new_id()
        if idem:
            inserted, existing = node.workflow_store.try_set_run_idempotency(idem, run_id)
            if not inserted:
                return {"run_id": existing, "deduplicated": True}
        # max concurrent runs (gateway.go:2783-2788)
        eff = node.configsvc.effective(org=wf.org_id)
        limit = int(((eff.config.get("rate_limits") or {}).get("concurrent_workflows")) or 0)
        if limit > 0 and node.workflow_store.count_active_runs(wf.org_id) >= limit:
            raise HTTPException(429, "too many concurrent workflow runs")
        run = WorkflowRun(
            id=run_id,
            workflow_id=wf_id,
            org_id=wf.org_id,
            team_id=wf.team_id,
            input=run_input,
            triggered_by=p.id,
            idempotency_key=idem,
            dry_run=bool(body.get("dry_run", False)),
            labels=dict(body.get("labels") or {}),
        )
        node.workflow_store.create_run(run)
        node.workflow.start_run(wf_id, run_id)
        node.drain()
        return {"run_id": run_id, "workflow_id": wf_id, "status": node.workflow_store.get_run(run_id).status}

    @api.get("/workflows/{wf_id}/runs")
    def list_wf_runs(wf_id: str, p: Principal = Depends(principal), limit: int = 50,
                     cursor: Optional[float] = None, status: str = ""):
        runs, next_cursor = node.workflow_store.list_runs(workflow_id=wf_id, status=status,
                                                          limit=limit, cursor=cursor)
        return {"items": [_run_summary(r) for r in runs], "next_cursor": next_cursor}

    # ------------------------------------------------------------------- runs
    @api.get("/workflow-runs")
    def list_runs(p: Principal = Depends(principal), workflow_id: str = "", status: str = "",
                  limit: int = 50, cursor: Optional[float] = None):
        runs, next_cursor = node.workflow_store.list_runs(workflow_id=workflow_id, status=status,
                                                          limit=limit, cursor=cursor)
        return {"items": [_run_summary(r) for r in runs], "next_cursor": next_cursor}

    @api.get("/workflow-runs/{run_id}")
    def get_run(run_id: str, p: Principal = Depends(principal)):
        try:
            return node.workflow_store.get_run(run_id).to_dict()
        except KeyError:
            raise HTTPException(404, "run not found")

    @api.delete("/workflow-runs/{run_id}")
    def delete_run(run_id: str, p: Principal = Depends(principal)):
        if not node.workflow_store.delete_run(run_id):
            raise HTTPException(404, "run not found")
        return {"deleted": run_id}

    @api.get("/workflow-runs/{run_id}/timeline")
    def run_timeline(run_id: str, p: Principal = Depends(principal)):
        return {"items": [e.to_dict() for e in node.workflow_store.get_timeline(run_id)]}

    @api.post("/workflow-runs/{run_id}/rerun")
    async def rerun(run_id: str, request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        try:
            new_run = node.workflow.rerun_from(run_id, body.get("step_id", ""),
                                               bool(body.get("dry_run", False)))
        except (KeyError, ValueError) as e:
            raise HTTPException(400, str(e))
        run = node.workflow_store.get_run(new_run)
        node.workflow.start_run(run.workflow_id, new_run)
        node.drain()
        return {"run_id": new_run, "rerun_of": run_id}

    @api.post("/workflows/{wf_id}/runs/{run_id}/cancel")
    def cancel_run(wf_id: str, run_id: str, p: Principal = Depends(principal)):
        try:
            node.workflow.cancel_run(run_id)
        except KeyError:
            raise HTTPException(404, "run not found")
        node.drain()
        return {"run_id": run_id, "status": "cancelled"}

    @api.post("/workflows/{wf_id}/runs/{run_id}/steps/{step_id}/approve")
    async def approve_wf_step(wf_id: str, run_id: str, step_id: str, request: Request,
                              p: Principal = Depends(admin)):
        body = await _json_body(request)
        approved = bool(body.get("approved", True))
        try:
            node.workflow.approve_step(run_id, step_id, approved)
        except (KeyError, ValueError) as e:
            raise HTTPException(409, str(e))
        node.drain()
        return {"run_id": run_id, "step_id": step_id, "approved": approved}

    # -------------------------------------------------------------- approvals
    @api.get("/approvals")
    def list_approvals(p: Principal = Depends(principal), limit: int = 50,
                       cursor: Optional[int] = None):
        """ListJobsByState(APPROVAL_REQUIRED) + safety summary, micros-cursor
        pagination (gateway.go:3644-3698)."""
        cursor = _normalize_micros(cursor) if cursor else None
        ids, next_cursor = node.job_store.list_jobs_by_state_page(
            JobState.APPROVAL_REQUIRED, cursor=cursor, limit=limit)
        items = []
        for jid in ids:
            meta = node.job_store.get_job_meta(jid)
            item = _job_summary(jid, meta)
            rec = node.job_store.get_safety_decision(jid)
            if rec is not None:
                item["safety"] = _safety_record_json(rec)
            items.append(item)
        return {"items": items, "next_cursor": next_cursor}

    @api.post("/approvals/{job_id}/approve")
    async def approve_job(job_id: str, request: Request, p: Principal = Depends(admin)):
        """Approval binding: snapshot + job-hash equality (gateway.go:3700-3835)."""
        body = await _json_body(request)
        state = node.job_store.get_state(job_id)
        if state != JobState.APPROVAL_REQUIRED:
            raise HTTPException(409, "job not awaiting approval")
        rec = node.job_store.get_safety_decision(job_id)
        if rec is None or not rec.policy_snapshot:
            raise HTTPException(409, "no recorded safety decision")
        snapshots = node.safety_kernel.list_snapshots()
        if rec.policy_snapshot not in snapshots:
            raise HTTPException(409, "policy snapshot changed since decision")
        req = node.job_store.get_job_request(job_id)
        if req is None:
            raise HTTPException(409, "job request not found")
        if job_hash(req) != rec.job_hash:
            raise HTTPException(409, "job hash mismatch")
        reason = (body.get("reason") or "").strip()
        note = (body.get("note") or "").strip()
        req.labels["approval_granted"] = "true"
        if reason:
            req.labels["approval_reason"] = reason
        if note:
            req.labels["approval_note"] = note
        req.labels[BUS_MSG_ID_LABEL] = f"approval:{new_id()}"
        node.job_store.set_job_request(job_id, req)
        node.job_store.set_approval_record(job_id, ApprovalRecord(
            approved_by=p.id, role=p.role, approved_at=node.clock.now_micros(),
            reason=reason, note=note, policy_snapshot=rec.policy_snapshot,
            job_hash=rec.job_hash, decision="approved",
        ))
        node.job_store.set_state(job_id, JobState.PENDING)
        node.submit_job(req, trace_id=node.job_store.get_job_meta(job_id).get("trace_id", ""))
        node.drain()
        return {"job_id": job_id, "state": str(node.job_store.get_state(job_id))}

    @api.post("/approvals/{job_id}/reject")
    async def reject_job(job_id: str, request: Request, p: Principal = Depends(admin)):
        body = await _json_body(request)
        state = node.job_store.get_state(job_id)
        if state != JobState.APPROVAL_REQUIRED:
            raise HTTPException(409, "job not awaiting approval")
        rec = node.job_store.get_safety_decision(job_id)
        node.job_store.set_approval_record(job_id, ApprovalRecord(
            approved_by=p.id, role=p.role, approved_at=node.clock.now_micros(),
            reason=(body.get("reason") or "").strip(),
            policy_snapshot=rec.policy_snapshot if rec else "",
            job_hash=rec.job_hash if rec else "", decision="rejected",
        ))
        node.job_store.set_state(job_id, JobState.DENIED)
        meta = node.job_store.get_job_meta(job_id)
        node.scheduler.emit_dlq(job_id, meta.get("topic", ""), JobStatus.DENIED,
                                "approval rejected", "approval_rejected")
        node.drain()
        return {"job_id": job_id, "state": str(node.job_store.get_state(job_id))}

    # ----------------------------------------------------------------- config
    @api.get("/config")
    def get_config(p: Principal = Depends(principal), scope: str = "system", id: str = "default"):
        doc = node.configsvc.get(scope, id)
        return {"scope": scope, "id": id, "config": doc or {}, "revision": node.configsvc.revision(scope, id)}

    @api.post("/config")
    async def set_config(request: Request, p: Principal = Depends(admin)):
        body = await _json_body(request)
        scope = body.get("scope") or "system"
        doc_id = body.get("id") or "default"
        config = body.get("config")
        if not isinstance(config, dict):
            raise HTTPException(400, "config object required")
        if body.get("merge"):
            rev = node.configsvc.patch(scope, doc_id, config)
        else:
            rev = node.configsvc.set(scope, doc_id, config)
        return {"scope": scope, "id": doc_id, "revision": rev}

    @api.get("/config/effective")
    def effective_config(p: Principal = Depends(principal), org: str = "", team: str = "",
                         workflow: str = "", step: str = ""):
        snap = node.configsvc.effective(org=org, team=team, workflow=workflow, step=step)
        return {"config": snap.config, "version": snap.version, "hash": snap.hash}

    # ---------------------------------------------------------------- schemas
    @api.post("/schemas")
    async def put_schema(request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        schema_id = (body.get("id") or "").strip()
        schema = body.get("schema")
        if not schema_id or not isinstance(schema, dict):
            raise HTTPException(400, "id and schema required")
        node.schemas.put(schema_id, schema)
        return {"id": schema_id}

    @api.get("/schemas")
    def list_schemas(p: Principal = Depends(principal)):
        return {"items": node.schemas.list()}

    @api.get("/schemas/{schema_id:path}")
    def get_schema(schema_id: str, p: Principal = Depends(principal)):
        schema = node.schemas.get(schema_id)
        if schema is None:
            raise HTTPException(404, "schema not found")
        return {"id": schema_id, "schema": schema}

    @api.delete("/schemas/{schema_id:path}")
    def delete_schema(schema_id: str, p: Principal = Depends(principal)):
        if not node.schemas.delete(schema_id):
            raise HTTPException(404, "schema not found")
        return {"deleted": schema_id}

    # ------------------------------------------------------------------ locks
    @api.get("/locks")
    def list_locks(p: Principal = Depends(principal)):
        return {
            "items": [
                {"resource": l.resource, "mode": l.mode, "owners": dict(l.owners),
                 "expires_at": l.expires_at}
                for l in node.locks.list()
            ]
        }

    @api.post("/locks/acquire")
    async def acquire_lock(request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        ok = node.locks.acquire(
            body.get("resource", ""), body.get("owner", p.id or "anonymous"),
            body.get("mode", "exclusive"), float(body.get("ttl_sec", 30)),
        )
        if not ok:
            raise HTTPException(409, "lock unavailable")
        return {"acquired": True}

    @api.post("/locks/release")
    async def release_lock(request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        ok = node.locks.release(body.get("resource", ""), body.get("owner", p.id or "anonymous"))
        return {"released": ok}

    @api.post("/locks/renew")
    async def renew_lock(request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        ok = node.locks.renew(body.get("resource", ""), body.get("owner", p.id or "anonymous"),
                              float(body.get("ttl_sec", 30)))
        if not ok:
            raise HTTPException(409, "lock not held")
        return {"renewed": True}

    # -------------------------------------------------------------------- DLQ
    @api.get("/dlq")
    def list_dlq(p: Principal = Depends(principal), limit: int = 100):
        entries, _ = node.dlq.list(limit=limit)
        return {"items": [e.to_dict() for e in entries]}

    @api.get("/dlq/page")
    def page_dlq(p: Principal = Depends(principal), limit: int = 50, cursor: Optional[int] = None):
        entries, next_cursor = node.dlq.list(limit=limit, cursor=cursor)
        return {"items": [e.to_dict() for e in entries], "next_cursor": next_cursor}

    @api.delete("/dlq/{job_id}")
    def delete_dlq(job_id: str, p: Principal = Depends(principal)):
        if not node.dlq.delete(job_id):
            raise HTTPException(404, "dlq entry not found")
        return {"deleted": job_id}

    @api.post("/dlq/{job_id}/retry")
    def retry_dlq(job_id: str, p: Principal = Depends(principal)):
        """gateway.go:3452-3553: new job `<id>-retry-<8>` w/ copied ctx."""
        entry = node.dlq.get(job_id)
        if entry is None:
            raise HTTPException(404, "dlq entry not found")
        orig = node.job_store.get_job_request(job_id)
        if orig is None:
            raise HTTPException(404, "original request not found")
        new_job_id = f"{job_id}-retry-{short_id(8)}"
        clone = JobRequest.decode(orig.encode())
        clone.job_id = new_job_id
        labels = {k: v for k, v in clone.labels.items() if k != BUS_MSG_ID_LABEL}
        labels.update(retry="true", dlq_entry=job_id, retry_of_job=job_id)
        labels[BUS_MSG_ID_LABEL] = f"dlqretry:{new_id()}"
        clone.labels = labels
        ctx = node.memory.get(f"ctx:{job_id}")
        node.job_store.set_state(new_job_id, JobState.PENDING)
        node.job_store.set_job_meta(new_job_id, topic=clone.topic, tenant=clone.tenant_id)
        node.submit_job(clone, context=ctx)
        node.drain()
        return {"job_id": new_job_id, "retry_of": job_id}

    # ----------------------------------------------------------------- policy
    @api.post("/policy/evaluate")
    @api.post("/policy/simulate")
    @api.post("/policy/explain")
    async def policy_check(request: Request, p: Principal = Depends(principal)):
        body = await _json_body(request)
        req = _policy_check_from_body(body, auth.resolve_tenant(p, body.get("tenant", "")))
        resp = node.safety_kernel.evaluate(req)
        out = resp.to_dict(camel=False)
        if request.url.path.endswith("/explain"):
            out["rules"] = node.safety_kernel.explain_rows(req)
        return out

    @api.get("/policy/snapshots")
    def policy_snapshots(p: Principal = Depends(principal)):
        return {"items": node.safety_kernel.list_snapshots(), "current": node.safety_kernel.snapshot}

    @api.get("/policy/rules")
    def policy_rules(p: Principal = Depends(principal)):
        policy = node.safety_kernel.current_policy()
        rules = policy.effective_rules() if policy else []
        return {
            "items": [
                {"id": r.id, "decision": r.decision, "reason": r.reason,
                 "match": {
                     "tenants": r.match.tenants, "topics": r.match.topics,
                     "capabilities": r.match.capabilities, "risk_tags": r.match.risk_tags,
                     "requires": r.match.requires, "labels": r.match.labels,
                 }}
                for r in rules
            ]
        }

    app.include_router(api)

    # policy-bundle studio + packs live in their own modules
    from .policy_bundles import make_policy_bundles_router
    from .packs import make_packs_router

    app.include_router(make_policy_bundles_router(node, principal, admin), prefix="/api/v1")
    app.include_router(make_packs_router(node, principal, admin), prefix="/api/v1")

    # ------------------------------------------------------------- WS stream
    # WebSocket/WebSocketDisconnect imported at module level: with
    # `from __future__ import annotations` a function-local import leaves
    # the stringified `WebSocket` annotation unresolvable and FastAPI
    # demoted the socket parameter to a required query field, closing every
    # connection with 1008 (found by the WS stream e2e test)
    @app.websocket("/api/v1/stream")
    async def stream(ws: WebSocket):
        pr = auth.authenticate({k: v for k, v in ws.headers.items()})
        if pr is None:
            await ws.close(code=4401)
            return
        # when the key rode in on the subprotocol (browser WebSocket API has
        # no headers), echo the protocol name back or the client's handshake
        # validation rejects the upgrade (gateway.go:2154-2185)
        offered = ws.headers.get("sec-websocket-protocol", "")
        sub = "cordum-api-key" if offered.split(",")[0].strip() == "cordum-api-key" else None
        await ws.accept(subprotocol=sub)
        queue: List[Any] = []
        ws_clients.append(queue)
        import asyncio

        try:
            while True:
                while queue:
                    await ws.send_json(queue.pop(0))
                await asyncio.sleep(0.05)
        except WebSocketDisconnect:
            pass
        finally:
            if queue in ws_clients:
                ws_clients.remove(queue)

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.get("/metrics")
    def metrics():
        from ..utils.metrics import Metrics as _M

        m = getattr(node, "metrics_exporter", None)
        if m is None:
            return Response(b"", media_type="text/plain")
        return Response(m.exposition(), media_type="text/plain; version=0.0.4")

    from .dashboard import add_dashboard

    add_dashboard(app)
    # extension-registered routes go last, like the reference's registrar
    # hook (they may shadow nothing; the compat table is already mounted)
    for e in exts:
        if isinstance(e, RouteRegistrar):
            e.register_routes(app)
    return app


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------


async def _json_body(request: Request) -> Dict[str, Any]:
    raw = await request.body()
    if not raw:
        return {}
    try:
        body = json.loads(raw)
    except ValueError:
        raise HTTPException(400, "invalid json")
    if not isinstance(body, dict):
        raise HTTPException(400, "object body required")
    return body


def _normalize_micros(ts: Optional[int]) -> int:
    """gateway.go:2407-2421 — accept seconds/millis/micros/nanos, return
    micros (upper-bound variant without the sub-unit rounding: callers here
    compare strictly so the +999_999 fill is immaterial for paging)."""
    if not ts or ts <= 0:
        return 0
    if ts < 1_000_000_000_000:          # seconds
        return ts * 1_000_000
    if ts < 1_000_000_000_000_000:      # millis
        return ts * 1_000
    if ts < 1_000_000_000_000_000_000:  # micros
        return ts
    return ts // 1_000                  # nanos


def _job_summary(job_id: str, meta: Dict[str, Any]) -> Dict[str, Any]:
    return {
        "id": job_id,
        "state": meta.get("state", ""),
        "topic": meta.get("topic", ""),
        "tenant": meta.get("tenant", ""),
        "team": meta.get("team", ""),
        "trace_id": meta.get("trace_id", ""),
        "updated_at": meta.get("updated_at", 0),
        "attempts": meta.get("attempts", 0),
    }


def _safety_record_json(rec) -> Dict[str, Any]:
    return {
        "decision": rec.decision,
        "reason": rec.reason,
        "rule_id": rec.rule_id,
        "policy_snapshot": rec.policy_snapshot,
        "approval_required": rec.approval_required,
        "approval_ref": rec.approval_ref,
        "job_hash": rec.job_hash,
        "checked_at": rec.checked_at,
        "constraints": rec.constraints.to_dict(camel=False) if rec.constraints else None,
        "remediations": [r.to_dict(camel=False) for r in rec.remediations],
    }


def _run_summary(run) -> Dict[str, Any]:
    return {
        "id": run.id,
        "workflow_id": run.workflow_id,
        "org_id": run.org_id,
        "status": run.status,
        "created_at": run.created_at,
        "updated_at": run.updated_at,
        "started_at": run.started_at,
        "completed_at": run.completed_at,
        "rerun_of": run.rerun_of,
        "dry_run": run.dry_run,
    }


def _job_detail(node: Node, job_id: str) -> Dict[str, Any]:
    """gateway.go:1011-1173 — the ~35-field job detail response."""
    meta = node.job_store.get_job_meta(job_id)
    rec = node.job_store.get_safety_decision(job_id)
    approval = node.job_store.get_approval_record(job_id)
    ctx_ptr = pointer_for_key(f"ctx:{job_id}")
    res_ptr = meta.get("result_ptr", "")

    def _inline(ptr):
        if not ptr:
            return None
        try:
            blob = node.memory.get_pointer(ptr)
        except ValueError:
            return None
        if blob is None:
            return None
        try:
            return json.loads(blob)
        except ValueError:
            return None

    req = node.job_store.get_job_request(job_id)
    labels = dict(req.labels) if req else {}
    workflow_id = (req.workflow_id if req else "") or labels.get("workflow_id", "")
    resp: Dict[str, Any] = {
        "id": job_id,
        "state": meta.get("state", ""),
        "trace_id": meta.get("trace_id", ""),
        "context_ptr": ctx_ptr,
        "context": _inline(ctx_ptr),
        "result_ptr": res_ptr,
        "result": _inline(res_ptr),
        "topic": meta.get("topic", ""),
        "tenant": meta.get("tenant", ""),
        "actor_id": meta.get("actor_id", ""),
        "actor_type": meta.get("actor_type", ""),
        "idempotency_key": meta.get("idempotency_key", ""),
        "capability": meta.get("capability", ""),
        "pack_id": meta.get("pack_id", ""),
        "risk_tags": meta.get("risk_tags", []),
        "requires": meta.get("requires", []),
        "attempts": meta.get("attempts", 0),
        "safety_decision": rec.decision if rec else "",
        "safety_reason": rec.reason if rec else "",
        "safety_rule_id": rec.rule_id if rec else "",
        "safety_snapshot": rec.policy_snapshot if rec else "",
        "safety_constraints": rec.constraints.to_dict(camel=False) if rec and rec.constraints else None,
        "safety_remediations": [r.to_dict(camel=False) for r in rec.remediations] if rec else [],
        "safety_job_hash": rec.job_hash if rec else "",
        "approval_required": rec.approval_required if rec else False,
        "approval_ref": rec.approval_ref if rec else "",
        "labels": labels,
        "workflow_id": workflow_id,
        "run_id": labels.get("run_id", ""),
        "step_id": labels.get("step_id", ""),
    }
    dlq_entry = node.dlq.get(job_id)
    if dlq_entry is not None:
        if dlq_entry.reason:
            resp["error_message"] = dlq_entry.reason
        if dlq_entry.status:
            resp["error_status"] = dlq_entry.status
        if dlq_entry.reason_code:
            resp["error_code"] = dlq_entry.reason_code
        if dlq_entry.last_state:
            resp["last_state"] = dlq_entry.last_state
        if dlq_entry.attempts:
            resp["attempts"] = dlq_entry.attempts
    if approval is not None:
        resp["approval_by"] = approval.approved_by
        resp["approval_role"] = approval.role
        resp["approval_at"] = approval.approved_at
        if approval.reason:
            resp["approval_reason"] = approval.reason
        if approval.note:
            resp["approval_note"] = approval.note
        resp["approval_policy_snapshot"] = approval.policy_snapshot
        resp["approval_job_hash"] = approval.job_hash
    return resp


def _policy_check_from_body(body: Dict[str, Any], tenant: str) -> PolicyCheckRequest:
    """gateway.go:322-370 buildPolicyCheckRequest."""
    meta_body = body.get("meta") or {}
    req = PolicyCheckRequest(
        job_id=body.get("job_id") or "",
        topic=body.get("topic") or "",
        tenant=body.get("tenant") or body.get("org_id") or tenant,
        principal_id=body.get("principal_id") or "",
        memory_id=body.get("memory_id") or "",
        labels=dict(body.get("labels") or {}),
        meta=JobMetadata(
            actor_id=meta_body.get("actor_id") or "",
            actor_type=parse_actor_type(meta_body.get("actor_type") or ""),
            idempotency_key=meta_body.get("idempotency_key") or "",
            capability=meta_body.get("capability") or "",
            risk_tags=list(meta_body.get("risk_tags") or []),
            requires=list(meta_body.get("requires") or []),
            pack_id=meta_body.get("pack_id") or "",
        ),
    )
    eff = body.get("effective_config")
    if eff is not None:
        req.effective_config = json.dumps(eff).encode()
    return req


def _packet_json(pkt: BusPacket) -> Dict[str, Any]:
    return pkt.to_dict()
