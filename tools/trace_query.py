"""Trace one TPC-H query's per-operator GPU time: python tools/trace_query.py <q> [sf]."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
os.environ["SAIL_TRACE"] = "1"

import torch

import sail_amd
from sail_amd.datagen.tpch import register_tpch
from sail_amd.datagen.tpch_queries import QUERIES


def main():
    q = int(sys.argv[1])
    sf = float(sys.argv[2]) if len(sys.argv) > 2 else 100.0
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    s = sail_amd.SessionContext(device=dev)
    register_tpch(s, sf=sf)
    s.sql(QUERIES[q]).collect()  # warmup
    t0 = time.time()
    s.sql(QUERIES[q]).collect()
    if dev == "cuda":
        torch.cuda.synchronize()
    print(f"q{q}: {(time.time()-t0)*1000:.0f}ms")
    ev = sorted(s.last_trace.events, key=lambda e: -e.self_ms)[:12]
    for e in ev:
        print(f"  {e.op:<16} {e.detail:<16} self={e.self_ms:8.1f}ms rows={e.rows}")


if __name__ == "__main__":
    main()
