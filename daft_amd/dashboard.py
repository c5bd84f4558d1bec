"""Embedded query dashboard (ref: /root/reference/src/daft-dashboard/ —
axum HTTP server with live query/operator views, fed by a dashboard
subscriber).  FastAPI app + in-memory state; launch with
`daft_amd.dashboard.serve(port)` or mount the `app` elsewhere."""
from __future__ import annotations

import threading
import time
from collections import deque
from typing import Optional

from .context import Subscriber, attach_subscriber


class DashboardState:
    def __init__(self, max_queries: int = 200):
        self.queries = deque(maxlen=max_queries)
        self.by_id = {}
        self._lock = threading.Lock()

    def start(self, query_id, plan):
        with self._lock:
            rec = {"id": query_id, "plan": plan, "status": "running",
                   "started_at": time.time(), "seconds": None,
                   "error": None, "operators": []}
            self.queries.appendleft(rec)
            self.by_id[query_id] = rec

    def end(self, query_id, seconds, error):
        with self._lock:
            rec = self.by_id.get(query_id)
            if rec:
                rec["status"] = "failed" if error else "done"
                rec["seconds"] = seconds
                rec["error"] = error

    def operator(self, query_id, name, rows_out, seconds):
        with self._lock:
            rec = self.by_id.get(query_id)
            if rec is not None:
                rec["operators"].append({
                    "name": name, "rows_out": rows_out, "seconds": seconds})


class DashboardSubscriber(Subscriber):
    def __init__(self, state: DashboardState):
        self.state = state

    def on_query_start(self, query_id, explain):
        self.state.start(query_id, explain)

    def on_query_end(self, query_id, seconds, error):
        self.state.end(query_id, seconds, error)

    def on_operator_end(self, query_id, node_id, name, rows_in, rows_out,
                        seconds):
        self.state.operator(query_id, name, rows_out, seconds)


_state = DashboardState()


def make_app(state: Optional[DashboardState] = None):
    from fastapi import FastAPI, HTTPException
    from fastapi.responses import HTMLResponse
    st = state or _state
    app = FastAPI(title="daft_amd dashboard")

    @app.get("/api/queries")
    def queries():
        return list(st.queries)

    @app.get("/api/queries/{qid}")
    def query(qid: str):
        rec = st.by_id.get(qid)
        if rec is None:
            raise HTTPException(404, "unknown query")
        return rec

    @app.get("/", response_class=HTMLResponse)
    def index():
        cells = []
        for q in list(st.queries):
            secs = "" if q["seconds"] is None else f"{q['seconds']:.3f}"
            cells.append(f"<tr><td>{q['id']}</td><td>{q['status']}</td>"
                         f"<td>{secs}</td></tr>")
        return ("<html><body><h2>daft_amd queries</h2>"
                "<table border=1><tr><th>id</th><th>status</th>"
                f"<th>seconds</th></tr>{''.join(cells)}</table>"
                "</body></html>")

    return app


def attach(state: Optional[DashboardState] = None) -> DashboardSubscriber:
    sub = DashboardSubscriber(state or _state)
    attach_subscriber(sub)
    return sub


def serve(port: int = 8238, host: str = "127.0.0.1",
          state: Optional[DashboardState] = None):
    """Attach the subscriber and serve the dashboard in a daemon thread."""
    import uvicorn
    attach(state)
    app = make_app(state)
    cfg = uvicorn.Config(app, host=host, port=port, log_level="warning")
    server = uvicorn.Server(cfg)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    return server
