"""Flagship benchmark: derived TPC-H, all 22 queries, synthetic data in HBM.

Contract (driver): `python bench.py --gpus N --steps K --warmup W` — launched
under torch.distributed.run for N>1 (one rank per GPU over RCCL). One step =
executing all 22 derived TPC-H queries at the configured scale factor. Data
is generated on-device (synthetic, TPC-H spec shaped — see datagen/tpch.py)
during setup; generation and planning are untimed, query execution is timed.

Rank 0 prints exactly one JSON line with the whole-job metric.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

# reduce HBM fragmentation for the large transient intermediates
os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--sf", type=float, default=float(os.environ.get("SAIL_BENCH_SF", "100")))
    ap.add_argument("--workload", default="tpch", choices=["tpch", "clickbench", "delta"])
    ap.add_argument("--rows", type=int, default=100_000_000, help="clickbench rows")
    ap.add_argument("--device", default=None)
    ap.add_argument("--queries", default=None, help="comma-separated subset, e.g. 1,6,13")
    ap.add_argument("--print-times", action="store_true")
    ap.add_argument("--scan", default=os.environ.get("SAIL_BENCH_SCAN", "parquet"),
                    choices=["parquet", "resident"],
                    help="parquet: timed steps scan+GPU-decode parquet from disk "
                         "(BASELINE config #2); resident: HBM-resident tables")
    args = ap.parse_args()

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)

    use_cuda = torch.cuda.is_available() if args.device is None else args.device.startswith("cuda")
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
    else:
        device = args.device or "cpu"

    dist = None
    if world > 1:
        import torch.distributed as tdist

        backend = "nccl" if use_cuda else "gloo"
        tdist.init_process_group(backend=backend)
        dist = tdist

    import sail_amd
    from sail_amd.exec.context import DistContext

    if args.workload == "clickbench":
        from sail_amd.datagen.clickbench import register_clickbench
        from sail_amd.datagen.clickbench_queries import QUERIES as CB
        QUERIES = {i: q for i, q in enumerate(CB)}
    elif args.workload == "delta":
        from sail_amd.datagen.delta_bench import DELTA_QUERIES as QUERIES
    else:
        from sail_amd.datagen.tpch_queries import QUERIES

    if use_cuda:
        from sail_amd.ops import kernels as K

        K.require()  # fail loudly if the HIP extension is missing on a GPU box

    session = sail_amd.SessionContext(device=device)
    if world > 1:
        session.dist = DistContext(dist, rank=rank, world=world, device=device)

    t0 = time.time()
    scan_info = None
    if args.workload == "clickbench":
        if args.scan == "parquet":
            from sail_amd.datagen.clickbench import register_clickbench_parquet

            scan_info = register_clickbench_parquet(
                session, rows=args.rows, device=device, rank=rank, world=world)
            print(f"# parquet shards: {scan_info['bytes']/1e9:.2f} GB in "
                  f"{scan_info['data_dir']}", file=sys.stderr)
        else:
            register_clickbench(session, rows=args.rows, device=device,
                                rank=rank, world=world)
    elif args.workload == "delta":
        from sail_amd.datagen.delta_bench import setup_delta_bench

        setup_delta_bench(session, sf=args.sf, device=device, rank=rank, world=world)
    elif args.scan == "parquet":
        # scan-inclusive: every timed query re-reads its columns from the
        # parquet shards on disk and decodes them on the GPU
        from sail_amd.datagen.tpch import register_tpch_parquet

        scan_info = register_tpch_parquet(session, sf=args.sf, device=device,
                                          rank=rank, world=world)
        print(f"# parquet shards: {scan_info['bytes']/1e9:.2f} GB in "
              f"{scan_info['data_dir']}", file=sys.stderr)
    else:
        from sail_amd.datagen.tpch import register_tpch

        register_tpch(session, sf=args.sf, device=device, rank=rank, world=world)
    if use_cuda:
        torch.cuda.synchronize()
    gen_s = time.time() - t0

    if args.queries:
        qids = [int(x) for x in args.queries.split(",")]
    elif args.workload == "clickbench":
        qids = list(range(len(QUERIES)))
    elif args.workload == "delta":
        qids = sorted(QUERIES.keys())
    else:
        qids = list(range(1, 23))
    if args.workload == "delta":
        from sail_amd.datagen.delta_bench import _delta_sql

        sqls = {q: _delta_sql(session, q) for q in qids}
    else:
        sqls = {q: QUERIES[q] for q in qids}
    plans = {}
    for q in qids:
        if args.workload == "delta":
            continue  # MERGE mutates state; plan fresh each step
        plans[q] = session.plan_sql(sqls[q])

    bench_tables = (["hits"] if args.workload == "clickbench" else
                    ["orders"] if args.workload == "delta" else
                    ["lineitem", "orders", "customer", "part", "partsupp", "supplier"])
    total_rows = sum(session.catalog.table_rows(t) or 0 for t in bench_tables)
    if dist is not None:
        tr = torch.tensor([total_rows], dtype=torch.int64,
                          device=device if use_cuda else "cpu")
        dist.all_reduce(tr)
        total_rows = int(tr.item())

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    progress = os.environ.get("SAIL_BENCH_PROGRESS") == "1"

    def one_step(collect_times=False):
        times = {}
        for q in qids:
            if progress and rank == 0:
                print(f"# start q{q}", file=sys.stderr, flush=True)
            tq = time.time()
            if q in plans:
                chunk = session.execute_plan(plans[q])
            else:
                chunk = session.sql(sqls[q]).collect_chunk()
            # result materialization to host is part of query completion
            if chunk.columns:
                _ = chunk.columns[0].data.cpu() if hasattr(chunk.columns[0], "data") else None
            if use_cuda:
                torch.cuda.synchronize()
            times[q] = time.time() - tq
        return times

    for _ in range(args.warmup):
        one_step()

    barrier_sync()
    t0 = time.time()
    qtimes = None
    for _ in range(args.steps):
        qtimes = one_step(collect_times=args.print_times)
    barrier_sync()
    elapsed = time.time() - t0

    # max over ranks
    if dist is not None:
        te = torch.tensor([elapsed], dtype=torch.float64,
                          device=device if use_cuda else "cpu")
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())

    per_step = elapsed / args.steps
    if rank == 0:
        if args.print_times and qtimes:
            for q in qids:
                print(f"# q{q}: {qtimes[q]*1000:.1f} ms", file=sys.stderr)
        print(f"# datagen: {gen_s:.1f}s, rows/GPU: {total_rows // max(world,1)}", file=sys.stderr)
        if args.workload == "clickbench":
            baseline = None
            metric = f"clickbench_{args.rows//1_000_000}m_total_s"
            model = "ClickBench (43 queries)"
        elif args.workload == "delta":
            baseline = None
            metric = f"delta_sf{args.sf:g}_scan_merge_s"
            model = "Delta Lake scan + MERGE INTO"
        else:
            baseline = 52.81 if abs(args.sf - 100.0) < 1e-6 else None
            metric = f"tpch_sf{args.sf:g}_total_s"
            model = "derived TPC-H (22 queries)"
        out = {
            "metric": metric,
            "value": round(per_step, 4),
            "unit": "s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(per_step * 1000, 2),
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": round(per_step / baseline, 4) if baseline else None,
            "dtype": "exact-int64+fp64",
            "data": "synthetic",
            "config": {
                "model": model,
                "sf": args.sf if args.workload == "tpch" else None,
                "rows": args.rows if args.workload == "clickbench" else None,
                "queries": len(qids),
                "parallelism": f"sharded dp{n_gpus}" if n_gpus > 1 else "single-gpu",
                "scan": ("parquet->GPU-decode->HBM (disk included in timed region)"
                         if scan_info is not None else "HBM-resident"),
                "parquet_bytes": scan_info["bytes"] if scan_info else None,
            },
        }
        print(json.dumps(out))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
