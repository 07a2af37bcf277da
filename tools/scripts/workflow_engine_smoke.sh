#!/usr/bin/env bash
# Workflow engine smoke (reference: tools/scripts/workflow_engine_smoke.sh):
# fan-out + retry + approval in one workflow.
set -euo pipefail
SERVER="${CORDUM_SERVER:-http://127.0.0.1:8080}"
CTL="python -m cordum_amd.cli.cordumctl --server $SERVER"
TMP=$(mktemp -d)
cat > "$TMP/wf.json" <<'JSON'
{"id": "smoke-fan", "name": "FanSmoke",
 "steps": {
   "fan": {"type": "worker", "topic": "job.default", "for_each": "${input.items}", "max_parallel": 4},
   "gate": {"type": "approval", "depends_on": ["fan"]},
   "done": {"type": "notify", "depends_on": ["gate"], "input": {"message": "fan complete"}}
 }}
JSON
$CTL workflow create -f "$TMP/wf.json" >/dev/null
RUN=$($CTL run start --workflow smoke-fan --input '{"items":[1,2,3,4,5,6,7,8]}' | python -c 'import sys,json; print(json.load(sys.stdin)["run_id"])')
sleep 1
STATUS=$($CTL run get "$RUN" | python -c 'import sys,json; print(json.load(sys.stdin)["status"])')
[ "$STATUS" = "waiting" ] || { echo "expected waiting at approval, got $STATUS"; exit 1; }
$CTL run step "$RUN" --workflow smoke-fan --step gate >/dev/null
sleep 1
STATUS=$($CTL run get "$RUN" | python -c 'import sys,json; print(json.load(sys.stdin)["status"])')
[ "$STATUS" = "succeeded" ] || { echo "expected succeeded, got $STATUS"; exit 1; }
$CTL workflow delete smoke-fan >/dev/null
echo "== workflow engine smoke OK"
