"""Operator traces for TPC-H queries in SCAN mode at a given SF.
Usage: python tools/trace_tpch_scan.py [sf] [q,q,...]"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sail_amd
from sail_amd.datagen.tpch import register_tpch_parquet
from sail_amd.datagen.tpch_queries import QUERIES


def main():
    sf = float(sys.argv[1]) if len(sys.argv) > 1 else 100.0
    qids = [int(x) for x in (sys.argv[2] if len(sys.argv) > 2
                             else "13,9,21").split(",")]
    dev = "cuda" if __import__("torch").cuda.is_available() else "cpu"
    s = sail_amd.SessionContext(device=dev)
    t0 = time.time()
    register_tpch_parquet(s, sf=sf, device=dev)
    print(f"# setup: {time.time() - t0:.1f}s", flush=True)
    for q in qids:
        s.sql(QUERIES[q]).collect_chunk()  # warm page cache
        s.conf["sail.trace"] = "true"
        t1 = time.time()
        df = s.sql(QUERIES[q])
        df.collect_chunk()
        dt = time.time() - t1
        s.conf["sail.trace"] = "false"
        print(f"===== q{q}: {dt*1000:.1f} ms")
        tr = getattr(s, "last_trace", None)
        if tr is not None:
            for e in tr.events:
                print(f"  {'  '*e.depth}{e.op:<22} {e.self_ms:9.2f}ms self "
                      f"rows={e.rows} {e.detail[:60]}")


if __name__ == "__main__":
    main()
