#!/usr/bin/env bash
# CLI smoke (reference: tools/scripts/cordumctl_smoke.sh): every top-level
# cordumctl command against a running node.
set -euo pipefail
SERVER="${CORDUM_SERVER:-http://127.0.0.1:8080}"
CTL="python -m cordum_amd.cli.cordumctl --server $SERVER"

$CTL status >/dev/null && echo "== status"
JOB=$($CTL job submit --prompt "cli smoke" --topic job.default | python -c 'import sys,json; print(json.load(sys.stdin)["job_id"])')
sleep 0.5
$CTL job status "$JOB" >/dev/null && echo "== job submit/status"
$CTL dlq list >/dev/null && echo "== dlq list"
$CTL approval list >/dev/null && echo "== approval list"
$CTL workflow list >/dev/null && echo "== workflow list"
$CTL run list >/dev/null && echo "== run list"
$CTL pack list >/dev/null && echo "== pack list"
echo "== cordumctl smoke OK"
