"""Trace ClickBench query ops: python tools/trace_cb.py <q> [rows]"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
os.environ["SAIL_TRACE"] = "1"

import torch

import sail_amd
from sail_amd.datagen.clickbench import register_clickbench
from sail_amd.datagen.clickbench_queries import QUERIES

q = int(sys.argv[1])
rows = int(sys.argv[2]) if len(sys.argv) > 2 else 100_000_000
dev = "cuda" if torch.cuda.is_available() else "cpu"
s = sail_amd.SessionContext(device=dev)
register_clickbench(s, rows=rows)
s.sql(QUERIES[q]).collect()
t0 = time.time()
s.sql(QUERIES[q]).collect()
if dev == "cuda":
    torch.cuda.synchronize()
print(f"q{q}: {(time.time()-t0)*1000:.0f}ms")
for e in sorted(s.last_trace.events, key=lambda e: -e.self_ms)[:8]:
    print(f"  {e.op:<16} {e.detail:<14} self={e.self_ms:8.1f}ms rows={e.rows}")
