"""Telemetry export: per-operator spans + metric instruments.

The reference exports fastrace spans and a YAML-declared metric registry
over OTLP (ref: crates/sail-telemetry/src/, metrics/data/registry.yaml;
TracingExec wraps every operator). The image has no network collector, so
the exporter writes OTLP/JSON (the `ResourceSpans`/`ResourceMetrics` wire
shapes) to a file — point SAIL_OTEL_FILE at a path (or set
`sail.telemetry.file`) and every traced query appends one line of spans
and one of metrics. A real OTLP/HTTP endpoint is a transport swap.
"""
from __future__ import annotations

import json
import os
import threading
import time
import uuid
from typing import Dict, List, Optional

#: instrument names mirror the reference's registry.yaml
METRIC_OUTPUT_ROWS = "execution.output_row_count"
METRIC_ELAPSED = "execution.elapsed_compute_time"
METRIC_OPERATORS = "execution.operator_count"
METRIC_QUERIES = "session.query_count"


class MetricsRegistry:
    """Process-wide counters/histograms (monotonic sums in OTLP terms)."""

    def __init__(self):
        self._lock = threading.Lock()
        self._sums: Dict[tuple, float] = {}

    def add(self, name: str, value: float, **attrs):
        key = (name, tuple(sorted(attrs.items())))
        with self._lock:
            self._sums[key] = self._sums.get(key, 0.0) + value

    def snapshot(self) -> List[dict]:
        with self._lock:
            items = list(self._sums.items())
        out = []
        now = int(time.time() * 1e9)
        for (name, attrs), val in items:
            out.append({
                "name": name,
                "sum": {"dataPoints": [{
                    "asDouble": val,
                    "timeUnixNano": str(now),
                    "attributes": [{"key": k,
                                    "value": {"stringValue": str(v)}}
                                   for k, v in attrs],
                }], "isMonotonic": True,
                    "aggregationTemporality": 2},
            })
        return out


_METRICS = MetricsRegistry()


def metrics() -> MetricsRegistry:
    return _METRICS


def _span_id() -> str:
    return uuid.uuid4().hex[:16]


def export_trace(trace, sql: Optional[str], path: str,
                 service: str = "sail-mi355x"):
    """One traced query -> OTLP/JSON ResourceSpans + ResourceMetrics lines.
    Operator spans nest by trace-event depth (the TracingExec shape)."""
    trace_id = uuid.uuid4().hex
    t_end = time.time()
    total_ms = sum(e.ms for e in trace.events if e.depth == 0) or \
        sum(e.self_ms for e in trace.events)
    t_start = t_end - total_ms / 1e3
    root = {
        "traceId": trace_id, "spanId": _span_id(), "name": "ExecutePlan",
        "kind": 1,
        "startTimeUnixNano": str(int(t_start * 1e9)),
        "endTimeUnixNano": str(int(t_end * 1e9)),
        "attributes": ([{"key": "sql",
                         "value": {"stringValue": (sql or "")[:2000]}}]),
    }
    spans = [root]
    parent_at: Dict[int, str] = {-1: root["spanId"]}
    cursor = t_start
    for e in trace.events:
        sid = _span_id()
        start = cursor
        spans.append({
            "traceId": trace_id, "spanId": sid,
            "parentSpanId": parent_at.get(e.depth - 1, root["spanId"]),
            "name": e.op, "kind": 1,
            "startTimeUnixNano": str(int(start * 1e9)),
            "endTimeUnixNano": str(int((start + e.ms / 1e3) * 1e9)),
            "attributes": [
                {"key": "rows", "value": {"intValue": str(e.rows)}},
                {"key": "detail", "value": {"stringValue": e.detail}},
                {"key": "self_ms", "value": {"doubleValue": e.self_ms}},
            ],
        })
        parent_at[e.depth] = sid
        _METRICS.add(METRIC_OUTPUT_ROWS, e.rows, operator=e.op)
        _METRICS.add(METRIC_ELAPSED, e.self_ms, operator=e.op)
        _METRICS.add(METRIC_OPERATORS, 1, operator=e.op)
    _METRICS.add(METRIC_QUERIES, 1)
    resource = {"attributes": [{"key": "service.name",
                                "value": {"stringValue": service}}]}
    line_spans = {"resourceSpans": [{
        "resource": resource,
        "scopeSpans": [{"scope": {"name": "sail_amd"}, "spans": spans}],
    }]}
    line_metrics = {"resourceMetrics": [{
        "resource": resource,
        "scopeMetrics": [{"scope": {"name": "sail_amd"},
                          "metrics": _METRICS.snapshot()}],
    }]}
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    with open(path, "a") as f:
        f.write(json.dumps(line_spans) + "\n")
        f.write(json.dumps(line_metrics) + "\n")


def maybe_export(session, trace, sql: Optional[str]):
    path = session.conf.get("sail.telemetry.file") or \
        os.environ.get("SAIL_OTEL_FILE", "")
    if path and trace is not None:
        export_trace(trace, sql, path)
