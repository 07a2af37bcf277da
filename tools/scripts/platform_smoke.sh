#!/usr/bin/env bash
# Black-box smoke against a running node (tools/scripts/platform_smoke.sh in
# the reference): workflow -> run -> verify -> delete, via cordumctl.
set -euo pipefail
SERVER="${CORDUM_SERVER:-http://127.0.0.1:8080}"
CTL="python -m cordum_amd.cli.cordumctl --server $SERVER"

$CTL status >/dev/null
echo "== status ok"

TMP=$(mktemp -d)
cat > "$TMP/wf.json" <<'JSON'
{"id": "smoke-workflow", "name": "Smoke",
 "steps": {"echo": {"type": "worker", "topic": "job.default",
                    "input": {"message": "${input.message}"}}}}
JSON
$CTL workflow create -f "$TMP/wf.json" >/dev/null
echo "== workflow created"

RUN_ID=$($CTL run start --workflow smoke-workflow --input '{"message":"smoke"}' | python -c 'import sys,json; print(json.load(sys.stdin)["run_id"])')
echo "== run started: $RUN_ID"
sleep 1
STATUS=$($CTL run get "$RUN_ID" | python -c 'import sys,json; print(json.load(sys.stdin)["status"])')
[ "$STATUS" = "succeeded" ] || { echo "run status: $STATUS (expected succeeded)"; exit 1; }
echo "== run succeeded"
$CTL run timeline "$RUN_ID" >/dev/null
$CTL workflow delete smoke-workflow >/dev/null
echo "== smoke OK"
