"""Observability subscribers (ref: /root/reference/daft/subscribers/ —
abc.py Subscriber, event_log.py JSONL sink; daft-context event dispatch).

Attach with daft_amd.attach_subscriber(...).  Events: query start/end,
optimization start/end, exec start/end, per-operator stats, heartbeats.
"""
from __future__ import annotations

import json
import os
import sys
import threading
import time
from typing import Optional

from ..context import Subscriber


class DebugSubscriber(Subscriber):
    """Print every event to stderr (ref: daft-context subscribers/debug.rs)."""

    def __init__(self, file=None):
        self.file = file or sys.stderr

    def _p(self, *args):
        print("[daft_amd]", *args, file=self.file, flush=True)

    def on_query_start(self, query_id, explain):
        self._p("query start", query_id)

    def on_query_end(self, query_id, seconds, error):
        self._p("query end", query_id, f"{seconds:.3f}s",
                f"error={error}" if error else "ok")

    def on_operator_end(self, query_id, node_id, name, rows_in, rows_out,
                        seconds):
        self._p(f"  op {name}: rows_in={rows_in} rows_out={rows_out} "
                f"{seconds * 1000:.1f}ms")


class EventLogSubscriber(Subscriber):
    """JSONL event log (ref: daft/subscribers/event_log.py:65; enable with
    DAFT_EVENT_LOG_DIR)."""

    def __init__(self, directory: Optional[str] = None):
        directory = directory or os.environ.get("DAFT_EVENT_LOG_DIR", ".")
        os.makedirs(directory, exist_ok=True)
        self.path = os.path.join(directory,
                                 f"daft_amd_events_{os.getpid()}.jsonl")
        self._lock = threading.Lock()

    def _w(self, event: str, **kw):
        rec = {"ts": time.time(), "event": event, **kw}
        with self._lock:
            with open(self.path, "a") as f:
                f.write(json.dumps(rec) + "\n")

    def on_query_start(self, query_id, explain):
        self._w("query_start", query_id=query_id, plan=explain)

    def on_query_end(self, query_id, seconds, error):
        self._w("query_end", query_id=query_id, seconds=seconds, error=error)

    def on_optimization_start(self, query_id):
        self._w("optimization_start", query_id=query_id)

    def on_optimization_end(self, query_id, seconds):
        self._w("optimization_end", query_id=query_id, seconds=seconds)

    def on_exec_start(self, query_id, node_names):
        self._w("exec_start", query_id=query_id, nodes=node_names)

    def on_operator_end(self, query_id, node_id, name, rows_in, rows_out,
                        seconds):
        self._w("operator_end", query_id=query_id, node_id=node_id,
                name=name, rows_in=rows_in, rows_out=rows_out,
                seconds=seconds)

    def on_exec_end(self, query_id):
        self._w("exec_end", query_id=query_id)

    def on_query_heartbeat(self, query_id):
        self._w("heartbeat", query_id=query_id)


class ProgressSubscriber(Subscriber):
    """Terminal progress lines per query (ref: daft/runners/progress_bar.py)."""

    def __init__(self, file=None):
        self.file = file or sys.stderr
        self._t0 = {}

    def on_query_start(self, query_id, explain):
        self._t0[query_id] = time.time()
        print(f"[daft_amd] running query {query_id} ...", file=self.file,
              flush=True)

    def on_query_end(self, query_id, seconds, error):
        status = "failed" if error else "done"
        print(f"[daft_amd] query {query_id} {status} in {seconds:.2f}s",
              file=self.file, flush=True)


import enum


class StatType(enum.Enum):
    """Operator stat kinds surfaced to subscribers (ref:
    daft/subscribers StatType)."""
    COUNT = "count"
    BYTES = "bytes"
    PERCENT = "percent"
    FLOAT = "float"
    DURATION = "duration"


def launch(detach: bool = False, port: int = 8238):
    """Launch the dashboard subscriber (ref: daft.subscribers.launch —
    starts the dashboard server and attaches its subscriber)."""
    from ..dashboard import serve
    return serve(port=port)
