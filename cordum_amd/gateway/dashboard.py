"""Minimal built-in dashboard: a single-page view over the compat API.

The reference ships a full React/Vite dashboard (18 pages) that talks to
`/api/v1`; that app works against this gateway unchanged.
This module serves a dependency-free status page at `/dashboard` for
deployments without the React bundle: jobs, runs, workers, DLQ, approvals,
policy snapshot — refreshed from the same endpoints the full dashboard uses.
"""

DASHBOARD_HTML = """<!doctype html>
<html><head><title>cordum-mi355x</title>
<style>
body{font-family:ui-monospace,monospace;background:#0b0e14;color:#d6deeb;margin:2rem}
h1{font-size:1.3rem} h2{font-size:1rem;color:#7fdbca;margin:1.2rem 0 .4rem}
table{border-collapse:collapse;width:100%;font-size:.8rem}
td,th{border:1px solid #1d2433;padding:.25rem .5rem;text-align:left}
th{background:#121826;color:#82aaff}
.ok{color:#7fdbca}.bad{color:#ef5350}.warn{color:#ffcb6b}
#status{margin:.5rem 0;font-size:.9rem}
input{background:#121826;color:#d6deeb;border:1px solid #1d2433;padding:.3rem}
</style></head><body>
<h1>cordum-mi355x control plane</h1>
<div>api key: <input id="key" placeholder="X-API-Key" size="24"></div>
<div id="status"></div>
<h2>workers</h2><div id="workers"></div>
<h2>recent jobs</h2><div id="jobs"></div>
<h2>workflow runs</h2><div id="runs"></div>
<h2>approvals pending</h2><div id="approvals"></div>
<h2>dead letter queue</h2><div id="dlq"></div>
<script>
const $=id=>document.getElementById(id);
async function get(path){
  const r=await fetch(path,{headers:{"X-API-Key":$("key").value||""}});
  if(!r.ok) throw new Error(r.status);
  return r.json();
}
function table(rows,cols){
  if(!rows||!rows.length) return "<i>none</i>";
  let h="<table><tr>"+cols.map(c=>`<th>${c}</th>`).join("")+"</tr>";
  for(const r of rows.slice(0,25))
    h+="<tr>"+cols.map(c=>`<td>${r[c]??""}</td>`).join("")+"</tr>";
  return h+"</table>";
}
async function refresh(){
  try{
    const s=await get("/api/v1/status");
    $("status").innerHTML=`<span class="ok">●</span> ok · uptime ${s.uptime_sec}s ·
      workers ${s.workers} · policy ${s.policy_snapshot}`;
    const w=await get("/api/v1/workers");
    $("workers").innerHTML=table(w.workers,["worker_id","pool","active_jobs","max_parallel_jobs","cpu_load","gpu_utilization"]);
    const j=await get("/api/v1/jobs?limit=25");
    $("jobs").innerHTML=table(j.items,["id","state","topic","tenant","attempts"]);
    const r=await get("/api/v1/workflow-runs?limit=25");
    $("runs").innerHTML=table(r.items,["id","workflow_id","status","rerun_of"]);
    const a=await get("/api/v1/approvals");
    $("approvals").innerHTML=table(a.items,["id","topic","tenant"]);
    const d=await get("/api/v1/dlq?limit=25");
    $("dlq").innerHTML=table(d.items,["job_id","topic","reason_code","reason","attempts"]);
  }catch(e){
    $("status").innerHTML=`<span class="bad">●</span> ${e}`;
  }
}
setInterval(refresh,3000); refresh();
</script></body></html>"""


def add_dashboard(app) -> None:
    from fastapi.responses import HTMLResponse

    @app.get("/dashboard", response_class=HTMLResponse)
    def dashboard():
        return DASHBOARD_HTML
