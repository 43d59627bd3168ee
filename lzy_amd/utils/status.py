"""Status endpoint — the single-node answer to the reference's web UI
(reference: site/ + frontend/ — task listing & admin over the cloud
fleet).  Serves JSON over HTTP:

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
            else:
                body = json.dumps({
                    "service": "lzy_amd",
                    "endpoints": ["/workflows", "/whiteboards", "/metrics"],
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
