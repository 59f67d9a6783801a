"""Per-operator execution tracing.

The analogue of the reference's TracingExec wrapper
(ref: crates/sail-telemetry/src/execution/physical_plan.rs:54): every
operator execution records wall time (with device sync), output rows, and
plan depth. Enable with SAIL_TRACE=1 or session.conf["sail.trace"]="true";
read session.last_trace / print_trace().
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch


@dataclass
class TraceEvent:
    depth: int
    op: str
    detail: str
    rows: int
    ms: float
    self_ms: float = 0.0


@dataclass
class Trace:
    events: List[TraceEvent] = field(default_factory=list)

    def render(self) -> str:
        out = []
        for e in self.events:
            pad = "  " * e.depth
            out.append(f"{pad}{e.op:<18} {e.self_ms:9.2f}ms self {e.ms:9.2f}ms total "
                       f"rows={e.rows} {e.detail}")
        return "\n".join(out)


class Tracer:
    def __init__(self, device):
        self.device = device
        self.trace = Trace()
        self.depth = 0
        self._child_ms: List[float] = [0.0]

    def _sync(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def wrap(self, op: str, detail: str, fn):
        self._sync()
        t0 = time.perf_counter()
        self.depth += 1
        self._child_ms.append(0.0)
        out = fn()
        self._sync()
        ms = (time.perf_counter() - t0) * 1000
        child = self._child_ms.pop()
        self.depth -= 1
        self._child_ms[-1] += ms
        rows = getattr(out, "num_rows", -1)
        self.trace.events.append(TraceEvent(self.depth, op, detail, rows, ms, ms - child))
        return out
