"""Status endpoint — the single-node answer to the reference's web UI
(reference: site/ + frontend/ — task listing & admin over the cloud
fleet).  Serves JSON over HTTP:

  /             HTML dashboard (tasks/whiteboards tables — the
                single-node answer to the reference's React SPA)
  /workflows    recent executions with task-state counts (from journals)
  /whiteboards  whiteboard index listing
  /metrics      Prometheus text exposition

Start with ``serve_status(storage_root, journal_dir, port=0)``.
"""
from __future__ import annotations

import glob
import json
import os
import tempfile
import threading
from typing import Optional


def _workflows_payload(journal_dir: str) -> list:
    from lzy_amd.sched import Journal

    out = []
    for path in sorted(
        glob.glob(os.path.join(journal_dir, "*.jsonl")),
        key=os.path.getmtime, reverse=True,
    )[:100]:
        states = Journal.replay(path)
        counts: dict = {}
        for s in states.values():
            counts[s] = counts.get(s, 0) + 1
        out.append({
            "execution_id": os.path.basename(path)[: -len(".jsonl")],
            "mtime": os.path.getmtime(path),
            "tasks": len(states),
            "states": counts,
        })
    return out


def _tasks_payload(journal_dir: str, execution_id: str) -> list:
    """Per-task detail for one execution (reference: the SPA's task
    table): task id, op name (from the 'scheduled' record detail),
    current state, and the failure detail if any."""
    path = os.path.join(journal_dir, f"{execution_id}.jsonl")
    if not os.path.exists(path) or os.path.basename(path) != f"{execution_id}.jsonl":
        return []
    tasks: dict = {}
    order: list = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            try:
                rec = json.loads(line)
            except ValueError:
                continue  # torn tail
            tid = rec.get("t")
            if tid not in tasks:
                tasks[tid] = {"task_id": tid, "name": "", "state": "",
                              "detail": ""}
                order.append(tid)
            t = tasks[tid]
            state = rec.get("s", "")
            detail = rec.get("d", "")
            if state == "scheduled" and detail:
                t["name"] = detail
            t["state"] = state
            if state in ("failed", "cancelled", "retry") and detail:
                t["detail"] = detail
    return [tasks[t] for t in order]


def _gpus_payload() -> list:
    """Node GPU inventory + live memory (the reference's 'pools' view)."""
    try:
        import torch

        if not torch.cuda.is_available():
            return []
        out = []
        for i in range(torch.cuda.device_count()):
            props = torch.cuda.get_device_properties(i)
            free, total = torch.cuda.mem_get_info(i)
            out.append({
                "index": i,
                "name": props.name,
                "total_gb": round(total / 1e9, 1),
                "used_gb": round((total - free) / 1e9, 1),
                "multi_processor_count": props.multi_processor_count,
            })
        return out
    except Exception:  # noqa: BLE001 - status must never crash
        return []


def _whiteboards_payload(storage_root: str) -> list:
    db = os.path.join(storage_root, "whiteboards.db")
    if not os.path.exists(db):
        return []
    from lzy_amd.whiteboards.index import WhiteboardIndexClient

    idx = WhiteboardIndexClient(db)
    return [
        {
            "id": m.id,
            "name": m.name,
            "tags": m.tags,
            "status": m.status,
            "created_at": m.created_at.isoformat(),
            "fields": sorted(m.fields.keys()),
        }
        for m in idx.query()
    ]


_DASHBOARD_HTML = """<!doctype html>
<html><head><meta charset="utf-8"><title>lzy-mi355x</title>
<style>
 body{font-family:system-ui,sans-serif;margin:2rem;background:#101418;color:#dde}
 h1{font-size:1.3rem} h2{font-size:1.05rem;margin-top:1.6rem}
 table{border-collapse:collapse;width:100%;font-size:.88rem}
 th,td{border:1px solid #2a3138;padding:.35rem .6rem;text-align:left}
 th{background:#1a2026} tr:nth-child(even){background:#161b20}
 .ok{color:#7c6} .bad{color:#e66} .run{color:#fc6}
 code{color:#9cf} a{color:#9cf;cursor:pointer}
 .cards{display:flex;gap:1rem;flex-wrap:wrap;margin:.8rem 0}
 .card{background:#161b20;border:1px solid #2a3138;border-radius:6px;
       padding:.6rem 1rem;min-width:8rem}
 .card b{display:block;font-size:1.3rem}
 #tasks{display:none}
</style></head><body>
<h1>lzy-mi355x &mdash; node status</h1>
<p>endpoints: <code>/workflows</code> <code>/workflows/&lt;id&gt;</code>
<code>/whiteboards</code> <code>/gpus</code> <code>/metrics</code>
&mdash; auto-refresh 5 s</p>
<div class="cards" id="cards"></div>
<h2>GPUs</h2><table id="gpu"><tr><th>#</th><th>name</th><th>CUs</th>
<th>HBM used / total (GB)</th></tr></table>
<h2>Executions <small>(click one for its tasks)</small></h2>
<table id="wf"><tr><th>execution</th><th>tasks</th>
<th>states</th><th>updated</th></tr></table>
<div id="tasks"><h2 id="tt"></h2><table id="tk"><tr><th>op</th>
<th>state</th><th>detail</th><th>task id</th></tr></table></div>
<h2>Whiteboards</h2><table id="wb"><tr><th>name</th><th>id</th>
<th>status</th><th>tags</th><th>fields</th><th>created</th></tr></table>
<script>
function cls(s){return s==='done'?'ok':(s==='failed'?'bad':'run')}
function clear(t){while(t.rows.length>1)t.deleteRow(1)}
function showTasks(id){fetch('/workflows/'+id).then(r=>r.json()).then(ts=>{
 document.getElementById('tasks').style.display='block';
 document.getElementById('tt').textContent='Tasks of '+id;
 const t=document.getElementById('tk');clear(t);
 ts.forEach(x=>{const r=t.insertRow();r.insertCell().textContent=x.name;
  const c=r.insertCell();c.textContent=x.state;c.className=cls(x.state);
  r.insertCell().textContent=x.detail;r.insertCell().textContent=x.task_id.slice(0,8)})})}
function refresh(){
fetch('/workflows').then(r=>r.json()).then(ws=>{const t=document.getElementById('wf');
 clear(t);
 ws.forEach(w=>{const r=t.insertRow();const a=document.createElement('a');
  a.textContent=w.execution_id;a.onclick=()=>showTasks(w.execution_id);
  r.insertCell().appendChild(a);
  r.insertCell().textContent=w.tasks;
  const c=r.insertCell();Object.entries(w.states).forEach(([s,n])=>{
   const sp=document.createElement('span');sp.className=cls(s);
   sp.textContent=s+':'+n+' ';c.appendChild(sp)});
  r.insertCell().textContent=new Date(w.mtime*1000).toISOString()})});
fetch('/whiteboards').then(r=>r.json()).then(ws=>{const t=document.getElementById('wb');
 clear(t);
 ws.forEach(w=>{const r=t.insertRow();r.insertCell().textContent=w.name;
  r.insertCell().textContent=w.id;r.insertCell().textContent=w.status;
  r.insertCell().textContent=w.tags.join(', ');
  r.insertCell().textContent=w.fields.join(', ');
  r.insertCell().textContent=w.created_at})});
fetch('/gpus').then(r=>r.json()).then(gs=>{const t=document.getElementById('gpu');
 clear(t);
 gs.forEach(g=>{const r=t.insertRow();r.insertCell().textContent=g.index;
  r.insertCell().textContent=g.name;
  r.insertCell().textContent=g.multi_processor_count;
  r.insertCell().textContent=g.used_gb+' / '+g.total_gb})});
fetch('/metrics').then(r=>r.text()).then(m=>{
 const want={'lzy_op_runs':'ops run','lzy_cache_hits':'cache hits',
  'lzy_transfers':'transfers','lzy_stream_plans':'stream plans',
  'lzy_op_failures':'op failures','lzy_task_retries':'task retries'};
 const vals={};m.split('\\n').forEach(l=>{const p=l.split(' ');
  const base=p[0].replace(/{.*/,'');
  if(want[base])vals[base]=(vals[base]||0)+parseFloat(p[1]||0)});
 const c=document.getElementById('cards');c.innerHTML='';
 Object.entries(want).forEach(([k,label])=>{const d=document.createElement('div');
  d.className='card';d.innerHTML='<b>'+(vals[k]||0)+'</b>'+label;
  c.appendChild(d)})});
}
refresh();setInterval(refresh,5000);
</script></body></html>
"""


def serve_status(
    storage_root: Optional[str] = None,
    journal_dir: Optional[str] = None,
    port: int = 0,
) -> int:
    """Start the status HTTP server; returns the bound port."""
    import http.server
    import socketserver

    from lzy_amd.utils.metrics import METRICS

    storage_root = storage_root or os.path.join(
        tempfile.gettempdir(), "lzy_amd_storage"
    )
    journal_dir = journal_dir or os.path.join(
        tempfile.gettempdir(), "lzy_amd_journal"
    )

    class Handler(http.server.BaseHTTPRequestHandler):
        def do_GET(self):  # noqa: N802
            if self.path.startswith("/workflows/"):
                eid = os.path.basename(self.path[len("/workflows/"):])
                body = json.dumps(
                    _tasks_payload(journal_dir, eid), indent=2
                ).encode()
                ctype = "application/json"
            elif self.path.startswith("/workflows"):
                body = json.dumps(_workflows_payload(journal_dir), indent=2).encode()
                ctype = "application/json"
            elif self.path.startswith("/gpus"):
                body = json.dumps(_gpus_payload(), indent=2).encode()
                ctype = "application/json"
            elif self.path.startswith("/whiteboards"):
                body = json.dumps(_whiteboards_payload(storage_root), indent=2).encode()
                ctype = "application/json"
            elif self.path.startswith("/metrics"):
                body = METRICS.render().encode()
                ctype = "text/plain; version=0.0.4"
            elif self.path in ("/", "/index.html"):
                body = _DASHBOARD_HTML.encode()
                ctype = "text/html"
            else:
                body = json.dumps({
                    "service": "lzy_amd",
                    "endpoints": ["/", "/workflows", "/workflows/<id>",
                                  "/whiteboards", "/gpus", "/metrics"],
                }).encode()
                ctype = "application/json"
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    srv = socketserver.TCPServer(("127.0.0.1", port), Handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    return srv.server_address[1]
