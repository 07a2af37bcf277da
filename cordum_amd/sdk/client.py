"""SDK client: REST wrapper over the gateway API.

Oracle: sdk/client/client.go (393 LoC) — workflows CRUD, runs (+dry-run,
idempotency header), timeline, approvals, DLQ, jobs submit/status/logs,
packs, artifacts. Python equivalent of the Go client; the wire surface is
the compat HTTP API, so either client works against either implementation.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import requests


class APIError(Exception):
    def __init__(self, status: int, body: str):
        super().__init__(f"HTTP {status}: {body[:200]}")
        self.status = status
        self.body = body


class Client:
    def __init__(self, base_url: str = "http://127.0.0.1:8080", api_key: str = "",
                 principal_id: str = "", role: str = "", timeout: float = 30.0,
                 session: Optional[requests.Session] = None):
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self.http = session or requests.Session()
        self.headers = {}
        if api_key:
            self.headers["X-API-Key"] = api_key
        if principal_id:
            self.headers["X-Principal-Id"] = principal_id
        if role:
            self.headers["X-Principal-Role"] = role

    def _req(self, method: str, path: str, **kw) -> Any:
        headers = dict(self.headers)
        headers.update(kw.pop("headers", {}))
        r = self.http.request(method, f"{self.base_url}{path}", headers=headers,
                              timeout=self.timeout, **kw)
        if r.status_code >= 400:
            raise APIError(r.status_code, r.text)
        if r.headers.get("content-type", "").startswith("application/json"):
            return r.json()
        return r.content

    # -- jobs -----------------------------------------------------------------
    def submit_job(self, prompt: str, topic: str = "job.default", **fields) -> Dict:
        body = {"prompt": prompt, "topic": topic, **fields}
        return self._req("POST", "/api/v1/jobs", json=body)

    def get_job(self, job_id: str) -> Dict:
        return self._req("GET", f"/api/v1/jobs/{job_id}")

    def list_jobs(self, **params) -> Dict:
        return self._req("GET", "/api/v1/jobs", params=params)

    def cancel_job(self, job_id: str) -> Dict:
        return self._req("POST", f"/api/v1/jobs/{job_id}/cancel")

    def remediate_job(self, job_id: str, remediation_id: str = "") -> Dict:
        return self._req("POST", f"/api/v1/jobs/{job_id}/remediate",
                         json={"remediation_id": remediation_id})

    # -- workflows / runs --------------------------------------------------------
    def create_workflow(self, workflow: Dict) -> Dict:
        return self._req("POST", "/api/v1/workflows", json=workflow)

    def get_workflow(self, wf_id: str) -> Dict:
        return self._req("GET", f"/api/v1/workflows/{wf_id}")

    def list_workflows(self) -> Dict:
        return self._req("GET", "/api/v1/workflows")

    def delete_workflow(self, wf_id: str) -> Dict:
        return self._req("DELETE", f"/api/v1/workflows/{wf_id}")

    def start_run(self, wf_id: str, input: Dict, dry_run: bool = False,
                  idempotency_key: str = "") -> Dict:
        headers = {}
        if idempotency_key:
            headers["X-Idempotency-Key"] = idempotency_key
        return self._req("POST", f"/api/v1/workflows/{wf_id}/runs",
                         json={"input": input, "dry_run": dry_run}, headers=headers)

    def get_run(self, run_id: str) -> Dict:
        return self._req("GET", f"/api/v1/workflow-runs/{run_id}")

    def list_runs(self, **params) -> Dict:
        return self._req("GET", "/api/v1/workflow-runs", params=params)

    def delete_run(self, run_id: str) -> Dict:
        return self._req("DELETE", f"/api/v1/workflow-runs/{run_id}")

    def run_timeline(self, run_id: str) -> Dict:
        return self._req("GET", f"/api/v1/workflow-runs/{run_id}/timeline")

    def rerun(self, run_id: str, step_id: str = "", dry_run: bool = False) -> Dict:
        return self._req("POST", f"/api/v1/workflow-runs/{run_id}/rerun",
                         json={"step_id": step_id, "dry_run": dry_run})

    def cancel_run(self, wf_id: str, run_id: str) -> Dict:
        return self._req("POST", f"/api/v1/workflows/{wf_id}/runs/{run_id}/cancel")

    def approve_step(self, wf_id: str, run_id: str, step_id: str, approved: bool = True) -> Dict:
        return self._req("POST", f"/api/v1/workflows/{wf_id}/runs/{run_id}/steps/{step_id}/approve",
                         json={"approved": approved})

    # -- approvals ----------------------------------------------------------------
    def list_approvals(self) -> Dict:
        return self._req("GET", "/api/v1/approvals")

    def approve_job(self, job_id: str, reason: str = "", note: str = "") -> Dict:
        return self._req("POST", f"/api/v1/approvals/{job_id}/approve",
                         json={"reason": reason, "note": note})

    def reject_job(self, job_id: str, reason: str = "") -> Dict:
        return self._req("POST", f"/api/v1/approvals/{job_id}/reject", json={"reason": reason})

    # -- DLQ --------------------------------------------------------------------
    def list_dlq(self) -> Dict:
        return self._req("GET", "/api/v1/dlq")

    def retry_dlq(self, job_id: str) -> Dict:
        return self._req("POST", f"/api/v1/dlq/{job_id}/retry")

    def delete_dlq(self, job_id: str) -> Dict:
        return self._req("DELETE", f"/api/v1/dlq/{job_id}")

    # -- packs ------------------------------------------------------------------
    def install_pack(self, archive: bytes) -> Dict:
        return self._req("POST", "/api/v1/packs/install", data=archive)

    def list_packs(self) -> Dict:
        return self._req("GET", "/api/v1/packs")

    def uninstall_pack(self, pack_id: str) -> Dict:
        return self._req("POST", f"/api/v1/packs/{pack_id}/uninstall")

    def verify_pack(self, pack_id: str) -> Dict:
        return self._req("POST", f"/api/v1/packs/{pack_id}/verify")

    # -- misc --------------------------------------------------------------------
    def status(self) -> Dict:
        return self._req("GET", "/api/v1/status")

    def workers(self) -> Dict:
        return self._req("GET", "/api/v1/workers")

    def artifacts_put(self, content: bytes, content_type: str = "application/octet-stream",
                      retention: str = "standard") -> Dict:
        return self._req("POST", "/api/v1/artifacts", data=content,
                         headers={"content-type": content_type, "x-retention": retention})

    def memory(self, ptr: str) -> Any:
        return self._req("GET", "/api/v1/memory", params={"ptr": ptr})

    def policy_evaluate(self, **body) -> Dict:
        return self._req("POST", "/api/v1/policy/evaluate", json=body)
