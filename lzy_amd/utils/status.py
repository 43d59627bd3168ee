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
 code{color:#9cf}
</style></head><body>
<h1>lzy-mi355x &mdash; node status</h1>
<p>endpoints: <code>/workflows</code> <code>/whiteboards</code>
<code>/metrics</code></p>
<h2>Executions</h2><table id="wf"><tr><th>execution</th><th>tasks</th>
<th>states</th><th>updated</th></tr></table>
<h2>Whiteboards</h2><table id="wb"><tr><th>name</th><th>id</th>
<th>status</th><th>tags</th><th>fields</th><th>created</th></tr></table>
<script>
function cls(s){return s==='done'?'ok':(s==='failed'?'bad':'run')}
fetch('/workflows').then(r=>r.json()).then(ws=>{const t=document.getElementById('wf');
 ws.forEach(w=>{const r=t.insertRow();r.insertCell().textContent=w.execution_id;
  r.insertCell().textContent=w.tasks;
  const c=r.insertCell();Object.entries(w.states).forEach(([s,n])=>{
   const sp=document.createElement('span');sp.className=cls(s);
   sp.textContent=s+':'+n+' ';c.appendChild(sp)});
  r.insertCell().textContent=new Date(w.mtime*1000).toISOString()})});
fetch('/whiteboards').then(r=>r.json()).then(ws=>{const t=document.getElementById('wb');
 ws.forEach(w=>{const r=t.insertRow();r.insertCell().textContent=w.name;
  r.insertCell().textContent=w.id;r.insertCell().textContent=w.status;
  r.insertCell().textContent=w.tags.join(', ');
  r.insertCell().textContent=w.fields.join(', ');
  r.insertCell().textContent=w.created_at})});
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
            if self.path.startswith("/workflows"):
                body = json.dumps(_workflows_payload(journal_dir), indent=2).encode()
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
                    "endpoints": ["/", "/workflows", "/whiteboards", "/metrics"],
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
