"""Run the derived TPC-H suite at any scale factor (the pysail example's
equivalent, ref: python/pysail/examples tpch runner).

    python examples/tpch_runner.py --sf 1 --queries 1,6,13 --scan parquet
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sail_amd
from sail_amd.datagen.tpch import register_tpch, register_tpch_parquet
from sail_amd.datagen.tpch_queries import QUERIES


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=0.1)
    ap.add_argument("--queries", default=",".join(str(q) for q in range(1, 23)))
    ap.add_argument("--scan", choices=["resident", "parquet"],
                    default="resident")
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    import torch

    dev = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    s = sail_amd.SessionContext(device=dev)
    t0 = time.time()
    if args.scan == "parquet":
        register_tpch_parquet(s, sf=args.sf, device=dev)
    else:
        register_tpch(s, sf=args.sf, device=dev)
    print(f"setup ({args.scan}, sf={args.sf}, {dev}): {time.time()-t0:.1f}s")
    total = 0.0
    for q in (int(x) for x in args.queries.split(",")):
        t1 = time.time()
        rows = s.sql(QUERIES[q]).collect()
        dt = time.time() - t1
        total += dt
        print(f"q{q:<3} {dt*1000:9.1f} ms   {len(rows)} rows")
    print(f"total {total:.3f}s")


if __name__ == "__main__":
    main()
