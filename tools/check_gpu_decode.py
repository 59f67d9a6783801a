"""Diagnostic: compare GPU parquet decode of the TPC-H shards against
pyarrow, column by column (counts, sums, min/max, random-sample equality).
Usage: python tools/check_gpu_decode.py [sf] [table ...]
"""
import sys

import numpy as np
import pyarrow.parquet as pq
import torch

sys.path.insert(0, ".")

from sail_amd.datagen.tpch import TpchGenerator, write_tpch_parquet  # noqa: E402
from sail_amd.datasource import gpu_parquet as G  # noqa: E402
from sail_amd.engine.column import StringColumn  # noqa: E402


def main():
    sf = float(sys.argv[1]) if len(sys.argv) > 1 else 1.0
    only = set(sys.argv[2:])
    dev = "cuda:0"
    gen = TpchGenerator(sf=sf, device=dev, seed=42, rank=0, world=1)
    tables = gen.generate_all()
    paths = write_tpch_parquet(tables, f"/tmp/sail_tpch_sf{sf:g}")
    del tables, gen
    torch.cuda.empty_cache()
    rng = np.random.default_rng(5)
    n_bad = 0
    for name, path in sorted(paths.items()):
        if only and name not in only:
            continue
        pf = pq.ParquetFile(path)
        for field in pf.schema_arrow:
            cn = field.name
            col = G.read_gpu([path], [(cn, None)], dev).columns[cn]
            torch.cuda.synchronize()
            host = pq.read_table(path, columns=[cn]).column(cn).combine_chunks()
            msgs = []
            if len(col) != len(host):
                msgs.append(f"len {len(col)} vs {len(host)}")
            n = len(host)
            idx = rng.integers(0, n, min(50_000, n)) if n else np.array([], int)
            if isinstance(col, StringColumn):
                hv = host.take(idx).to_pylist() if n else []
                gcol = col.gather(torch.from_numpy(idx).to(dev))
                gv = gcol.to_pylist()
                nbad = sum(1 for a, b in zip(gv, hv) if a != b)
                if nbad:
                    first = next((i, a, b) for i, (a, b)
                                 in enumerate(zip(gv, hv)) if a != b)
                    msgs.append(f"{nbad}/{len(idx)} sample mismatches, "
                                f"first={first}")
            else:
                hnp = host.to_numpy(zero_copy_only=False)
                if str(field.type).startswith("decimal"):
                    hscaled = np.array([None if v is None else int(
                        (v.scaleb(field.type.scale))) for v in host.to_pylist()],
                        dtype=np.int64)
                    hnp = hscaled
                elif str(field.type) == "date32[day]":
                    hnp = host.cast("int32").to_numpy(zero_copy_only=False)
                gd = col.data.cpu().numpy()
                if gd.dtype != hnp.dtype:
                    hnp = hnp.astype(gd.dtype)
                if len(gd) == len(hnp):
                    with np.errstate(over="ignore"):
                        if not np.array_equal(gd, hnp):
                            diff = np.nonzero(gd != hnp)[0]
                            i0 = int(diff[0])
                            msgs.append(
                                f"{len(diff)} mismatches, first at {i0}: "
                                f"gpu={gd[i0]} host={hnp[i0]}")
            if msgs:
                n_bad += 1
                print(f"BAD  {name}.{cn}: {'; '.join(msgs)}", flush=True)
            else:
                print(f"ok   {name}.{cn} ({n} rows)", flush=True)
    print(f"DONE bad_columns={n_bad}")


if __name__ == "__main__":
    main()
