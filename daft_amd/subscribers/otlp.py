"""OTLP span export (ref: /root/reference/src/common/tracing/src/
lib.rs:123-197 — the reference wires tracing_opentelemetry + OTLP
exporters enabled via env endpoint and flushes at query end).

Two sinks over the same span model:

- OTLPFileSpanExporter: OTLP/JSON `resourceSpans` documents appended to a
  JSONL file (one export batch per query end) — drop-in for offline
  environments; any OTLP collector ingests the same payload shape.
- OTLPHttpSpanExporter: POSTs the identical payload to an OTLP/HTTP
  endpoint (`v1/traces`), enabled by DAFT_AMD_OTLP_ENDPOINT.

Spans: one root span per query, child spans for the optimization phase
and each operator (wall-clock from the engine's per-op runtime stats).
"""
from __future__ import annotations

import json
import os
import time
import uuid
from typing import Dict, List, Optional

from ..context import Subscriber

_NS = 1_000_000_000


def _hex(nbytes: int) -> str:
    return uuid.uuid4().hex[: nbytes * 2]


class _SpanBuffer(Subscriber):
    """Collects per-query spans; subclasses implement export(payload)."""

    def __init__(self, service_name: str = "daft_amd"):
        self.service_name = service_name
        self._q: Dict[str, dict] = {}

    # -- event bus ---------------------------------------------------------
    def on_query_start(self, query_id, explain):
        self._q[query_id] = {
            "trace_id": _hex(16),
            "root": _hex(8),
            "t0": time.time_ns(),
            "spans": [],
            "explain": explain[:2000] if explain else "",
        }

    def on_optimization_start(self, query_id):
        q = self._q.get(query_id)
        if q is not None:
            q["opt_t0"] = time.time_ns()

    def on_optimization_end(self, query_id, seconds):
        q = self._q.get(query_id)
        if q is None:
            return
        t0 = q.pop("opt_t0", time.time_ns() - int(seconds * _NS))
        q["spans"].append(("optimize", t0, t0 + int(seconds * _NS), {}))

    def on_operator_end(self, query_id, node_id, name, rows, batches,
                        seconds):
        q = self._q.get(query_id)
        if q is None:
            return
        end = time.time_ns()
        q["spans"].append((name, end - int(seconds * _NS), end,
                           {"rows": rows, "batches": batches,
                            "node_id": node_id}))

    def on_query_end(self, query_id, seconds, error=None):
        q = self._q.pop(query_id, None)
        if q is None:
            return
        end = time.time_ns()
        spans = [{
            "traceId": q["trace_id"], "spanId": q["root"],
            "name": "query", "kind": 1,
            "startTimeUnixNano": str(q["t0"]),
            "endTimeUnixNano": str(end),
            "status": {"code": 2 if error else 1,
                       **({"message": str(error)[:500]} if error else {})},
            "attributes": [
                {"key": "daft.query_id",
                 "value": {"stringValue": query_id}},
            ],
        }]
        for name, t0, t1, attrs in q["spans"]:
            spans.append({
                "traceId": q["trace_id"], "spanId": _hex(8),
                "parentSpanId": q["root"], "name": name, "kind": 1,
                "startTimeUnixNano": str(t0),
                "endTimeUnixNano": str(t1),
                "attributes": [
                    {"key": f"daft.{k}",
                     "value": {"intValue": str(v)} if isinstance(v, int)
                     else {"stringValue": str(v)}}
                    for k, v in attrs.items()],
            })
        payload = {"resourceSpans": [{
            "resource": {"attributes": [
                {"key": "service.name",
                 "value": {"stringValue": self.service_name}}]},
            "scopeSpans": [{
                "scope": {"name": "daft_amd.tracing"},
                "spans": spans}],
        }]}
        self.export(payload)

    def export(self, payload: dict) -> None:  # pragma: no cover
        raise NotImplementedError


class OTLPFileSpanExporter(_SpanBuffer):
    def __init__(self, path: str, service_name: str = "daft_amd"):
        super().__init__(service_name)
        self.path = path

    def export(self, payload):
        with open(self.path, "a") as f:
            f.write(json.dumps(payload) + "\n")


class OTLPHttpSpanExporter(_SpanBuffer):
    def __init__(self, endpoint: Optional[str] = None,
                 service_name: str = "daft_amd", timeout: float = 5.0):
        super().__init__(service_name)
        self.endpoint = (endpoint or
                         os.environ.get("DAFT_AMD_OTLP_ENDPOINT", "")) \
            .rstrip("/")
        self.timeout = timeout

    def export(self, payload):
        if not self.endpoint:
            return
        import requests
        try:
            requests.post(f"{self.endpoint}/v1/traces", json=payload,
                          timeout=self.timeout)
        except Exception:
            pass      # tracing must never fail the query
