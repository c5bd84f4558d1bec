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
        return _UI_HTML

    return app


# self-contained live UI (ref: the reference ships a web app in
# daft-dashboard; this is the same query/operator browser as one page —
# polls /api/queries, click-through to per-operator stats and the plan)
_UI_HTML = """<!doctype html><html><head><title>daft_amd dashboard</title>
<style>
 body{font-family:ui-monospace,monospace;margin:1.5rem;background:#111;
      color:#ddd}
 h1{font-size:1.2rem} a{color:#7ab8ff;cursor:pointer}
 table{border-collapse:collapse;margin-top:.6rem;width:100%}
 th,td{border:1px solid #333;padding:.25rem .6rem;text-align:left;
       font-size:.85rem}
 th{background:#1c1c1c} tr:hover{background:#191919}
 .done{color:#7dd87d}.failed{color:#ff7a7a}.running{color:#ffd37a}
 #detail{margin-top:1rem;border-top:1px solid #333;padding-top:.8rem}
 pre{background:#181818;padding:.6rem;overflow-x:auto;font-size:.78rem}
 .bar{background:#2a6;display:inline-block;height:.6rem}
</style></head><body>
<h1>daft_amd — query dashboard</h1>
<div id="list"></div><div id="detail"></div>
<script>
async function refresh(){
  const qs = await (await fetch('/api/queries')).json();
  let h = '<table><tr><th>query</th><th>status</th><th>seconds</th>'+
          '<th>operators</th></tr>';
  for (const q of qs){
    h += `<tr><td><a onclick="show('${q.id}')">${q.id}</a></td>`+
         `<td class="${q.status}">${q.status}</td>`+
         `<td>${q.seconds==null?'':q.seconds.toFixed(3)}</td>`+
         `<td>${(q.operators||[]).length}</td></tr>`;
  }
  document.getElementById('list').innerHTML = h + '</table>';
}
async function show(id){
  const q = await (await fetch('/api/queries/'+id)).json();
  const ops = q.operators || [];
  const tmax = Math.max(1e-9, ...ops.map(o=>o.seconds));
  let h = `<h1>${q.id} — ${q.status}`+
          (q.seconds!=null?` (${q.seconds.toFixed(3)}s)`:``)+`</h1>`;
  if (q.error) h += `<pre class="failed">${q.error}</pre>`;
  h += '<table><tr><th>operator</th><th>rows out</th><th>seconds</th>'+
       '<th></th></tr>';
  for (const o of ops){
    const w = Math.round(160*o.seconds/tmax);
    h += `<tr><td>${o.name}</td><td>${o.rows_out}</td>`+
         `<td>${o.seconds.toFixed(4)}</td>`+
         `<td><span class="bar" style="width:${w}px"></span></td></tr>`;
  }
  h += '</table><h1>plan</h1><pre>'+
       (q.plan||'').replace(/</g,'&lt;')+'</pre>';
  document.getElementById('detail').innerHTML = h;
}
refresh(); setInterval(refresh, 2000);
</script></body></html>"""


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
