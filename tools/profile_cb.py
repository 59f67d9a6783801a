"""Operator trace for clickbench scan-mode queries."""
import os
import sys

sys.path.insert(0, ".")
os.environ["SAIL_TRACE"] = "1"

import sail_amd  # noqa: E402
from sail_amd.datagen.clickbench import register_clickbench_parquet  # noqa: E402
from sail_amd.datagen.clickbench_queries import QUERIES  # noqa: E402


def main():
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000_000
    qids = [int(x) for x in (sys.argv[2].split(",") if len(sys.argv) > 2 else ["0"])]
    import torch

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    s = sail_amd.SessionContext(device=dev)
    register_clickbench_parquet(s, rows=rows, device=dev)
    for q in qids:
        s.sql(QUERIES[q]).collect()
    for q in qids:
        s.sql(QUERIES[q]).collect()
        print(f"===== q{q}")
        print(s.last_trace.render())


if __name__ == "__main__":
    main()
