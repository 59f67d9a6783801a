"""Run one SQL string against synthetic hits: python tools/run_sql.py ROWS 'SQL'."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

import sail_amd
from sail_amd.datagen.clickbench import register_clickbench

rows = int(sys.argv[1])
sql = sys.argv[2]
dev = "cuda" if torch.cuda.is_available() else "cpu"
s = sail_amd.SessionContext(device=dev)
register_clickbench(s, rows=rows)
t0 = time.time()
out = s.sql(sql).collect()
if dev == "cuda":
    torch.cuda.synchronize()
print(f"ok {len(out)} rows {(time.time()-t0)*1000:.0f}ms :: {out[:3]}")
