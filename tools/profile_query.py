"""cProfile one query end-to-end on device: python tools/profile_query.py <q> [sf]."""
import cProfile
import io
import os
import pstats
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import sail_amd
from sail_amd.datagen.tpch import register_tpch
from sail_amd.datagen.tpch_queries import QUERIES

q = int(sys.argv[1])
sf = float(sys.argv[2]) if len(sys.argv) > 2 else 100.0
dev = "cuda" if torch.cuda.is_available() else "cpu"
s = sail_amd.SessionContext(device=dev)
register_tpch(s, sf=sf)
s.sql(QUERIES[q]).collect()
if dev == "cuda":
    torch.cuda.synchronize()
pr = cProfile.Profile()
t0 = time.time()
pr.enable()
s.sql(QUERIES[q]).collect()
if dev == "cuda":
    torch.cuda.synchronize()
pr.disable()
print(f"q{q}: {(time.time()-t0)*1000:.0f}ms")
st = io.StringIO()
pstats.Stats(pr, stream=st).sort_stats("cumulative").print_stats(28)
print(st.getvalue())
