"""Run one TPC-H query N times (for rocprofv3 kernel attribution):
python tools/run_query_n.py <q> [sf] [reps]"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import sail_amd
from sail_amd.datagen.tpch import register_tpch
from sail_amd.datagen.tpch_queries import QUERIES

q = int(sys.argv[1])
sf = float(sys.argv[2]) if len(sys.argv) > 2 else 100.0
reps = int(sys.argv[3]) if len(sys.argv) > 3 else 3
dev = "cuda" if torch.cuda.is_available() else "cpu"
s = sail_amd.SessionContext(device=dev)
register_tpch(s, sf=sf)
s.sql(QUERIES[q]).collect()
if dev == "cuda":
    torch.cuda.synchronize()
t0 = time.time()
for _ in range(reps):
    s.sql(QUERIES[q]).collect()
if dev == "cuda":
    torch.cuda.synchronize()
print(f"q{q}: {(time.time()-t0)*1000/reps:.1f}ms/run x{reps}")
