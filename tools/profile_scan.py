"""Per-operator timing of scan-mode TPC-H queries (SAIL_TRACE-based).
Usage: python tools/profile_scan.py <sf> <q1,q2,...>"""
import os
import sys

sys.path.insert(0, ".")
os.environ["SAIL_TRACE"] = "1"

import sail_amd  # noqa: E402
from sail_amd.datagen.tpch import register_tpch_parquet  # noqa: E402
from sail_amd.datagen.tpch_queries import QUERIES  # noqa: E402


def main():
    sf = float(sys.argv[1]) if len(sys.argv) > 1 else 1.0
    qids = [int(x) for x in (sys.argv[2].split(",") if len(sys.argv) > 2
                             else ["1", "6"])]
    import torch

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    s = sail_amd.SessionContext(device=dev)
    register_tpch_parquet(s, sf=sf, device=dev)
    for q in qids:
        s.sql(QUERIES[q]).collect()  # warm (page cache, indexes)
    for q in qids:
        s.sql(QUERIES[q]).collect()
        print(f"===== q{q}")
        print(s.last_trace.render())


if __name__ == "__main__":
    main()
