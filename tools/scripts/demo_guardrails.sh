#!/usr/bin/env bash
# Guardrails demo: install the pack, submit a guarded job, approve it.
set -euo pipefail
SERVER="${CORDUM_SERVER:-http://127.0.0.1:8080}"
CTL="python -m cordum_amd.cli.cordumctl --server $SERVER"
REPO_ROOT="$(cd "$(dirname "$0")/../.." && pwd)"

$CTL pack install -f "$REPO_ROOT/examples/demo-guardrails" >/dev/null
echo "== demo-guardrails installed"
JOB=$($CTL job submit --prompt "deploy the thing" --topic job.guarded | python -c 'import sys,json; print(json.load(sys.stdin)["job_id"])')
sleep 1
STATE=$($CTL job status "$JOB" | python -c 'import sys,json; print(json.load(sys.stdin)["state"])')
echo "== submitted $JOB state=$STATE"
