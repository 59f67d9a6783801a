"""Run ClickBench queries sequentially, printing index before each (crash isolation)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import sail_amd
from sail_amd.datagen.clickbench import register_clickbench
from sail_amd.datagen.clickbench_queries import QUERIES

rows = int(sys.argv[1]) if len(sys.argv) > 1 else 20_000_000
only = [int(x) for x in sys.argv[2].split(",")] if len(sys.argv) > 2 else range(len(QUERIES))
dev = "cuda" if torch.cuda.is_available() else "cpu"
s = sail_amd.SessionContext(device=dev)
register_clickbench(s, rows=rows)
total = 0.0
for i in only:
    print(f"Q{i}...", flush=True)
    t0 = time.time()
    s.sql(QUERIES[i]).collect()
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.time() - t0
    total += dt
    print(f"Q{i}: {dt*1000:.0f}ms", flush=True)
print(f"total {total:.2f}s @ {rows} rows")
