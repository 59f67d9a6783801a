"""Phase breakdown of the delta bench (BASELINE config #5) on one GPU:
setup / scan-agg / merge, with IO sub-phases (parquet read, arrow->device,
rewrite encode) timed via wrappers. Run: python tools/delta_phase_bench.py [sf]"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import sail_amd
from sail_amd.datagen import delta_bench
from sail_amd.datasource import delta, parquet_io, arrow_io

sf = float(sys.argv[1]) if len(sys.argv) > 1 else 10.0
dev = "cuda" if torch.cuda.is_available() else "cpu"

acc = {}


def wrap(mod, name, label):
    orig = getattr(mod, name)

    def timed(*a, **k):
        t0 = time.time()
        out = orig(*a, **k)
        acc[label] = acc.get(label, 0.0) + (time.time() - t0)
        return out

    setattr(mod, name, timed)


wrap(parquet_io, "read", "parquet_read")
wrap(arrow_io, "arrow_to_table", "arrow_to_device")
wrap(arrow_io, "chunk_to_arrow", "chunk_to_arrow")
wrap(delta, "_write_parts", "part_write")
# delta.read calls parquet_io.read through its module import
delta.parquet_io = parquet_io if hasattr(delta, "parquet_io") else None

s = sail_amd.SessionContext(device=dev)
t0 = time.time()
delta_bench.setup_delta_bench(s, sf=sf, device=dev)
if dev == "cuda":
    torch.cuda.synchronize()
print(f"setup: {time.time()-t0:.2f}s  io={dict((k, round(v,2)) for k,v in acc.items())}")

for step in range(2):
    acc.clear()
    t0 = time.time()
    s.sql(delta_bench._delta_sql(s, 1)).collect_chunk()
    if dev == "cuda":
        torch.cuda.synchronize()
    t1 = time.time()
    print(f"step{step} scan_agg: {t1-t0:.2f}s  io={dict((k, round(v,2)) for k,v in acc.items())}")
    acc.clear()
    s.sql(delta_bench._delta_sql(s, 2)).collect_chunk()
    if dev == "cuda":
        torch.cuda.synchronize()
    print(f"step{step} merge:    {time.time()-t1:.2f}s  io={dict((k, round(v,2)) for k,v in acc.items())}")
