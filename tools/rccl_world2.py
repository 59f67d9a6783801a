"""World=2 over RCCL on ONE GPU: shakes out device-tensor collective
ordering/dtype/stream bugs in the SPMD exchanges before any 8-GPU run
(VERDICT r1 item 2). Run under torchrun --nproc-per-node 2; both ranks pin
cuda:0. Forces the shuffle paths (sort/distinct/window + hash shuffle) at
tiny thresholds and checks all 22 TPC-H results against a single-process
truth computed on rank 0.
"""
import os
import sys

sys.path.insert(0, ".")
os.environ.setdefault("SAIL_DIST_SORT_MIN_ROWS", "1")
os.environ.setdefault("SAIL_DIST_DISTINCT_MIN_ROWS", "1")
os.environ.setdefault("SAIL_EXEC_BROADCAST_THRESHOLD_BYTES", "65536")
os.environ.setdefault("SAIL_EXEC_AGG_SHUFFLE_THRESHOLD_GROUPS", "64")


def main():
    import torch
    import torch.distributed as dist

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(0)  # both ranks share GPU 0 deliberately
    dist.init_process_group("nccl", rank=rank, world_size=world)

    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator, register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.exec.context import DistContext

    sf = float(os.environ.get("SAIL_W2_SF", "0.1"))
    s = sail_amd.SessionContext(device="cuda:0")
    s.dist = DistContext(dist, rank=rank, world=world, device="cuda:0")
    register_tpch(s, sf=sf, rank=rank, world=world)

    results = {}
    for q in range(1, 23):
        results[q] = s.sql(QUERIES[q]).collect()
    dist.barrier()
    torch.cuda.synchronize()

    if rank == 0:
        # single-process truth: union of both shards
        from sail_amd.engine.column import Table
        from sail_amd.engine.executor import concat_columns

        single = sail_amd.SessionContext(device="cuda:0")
        shards = [TpchGenerator(sf=sf, device="cuda:0", rank=r,
                                world=world).generate_all()
                  for r in range(world)]
        for name in shards[0]:
            if name in ("region", "nation"):
                single.catalog.register_table(name, shards[0][name])
                continue
            cols = {cn: concat_columns([shards[r][name].columns[cn]
                                        for r in range(world)])
                    for cn in shards[0][name].columns}
            single.catalog.register_table(name, Table(cols))
        bad = []
        for q in range(1, 23):
            want = single.sql(QUERIES[q]).collect()
            got = results[q]
            ok = len(got) == len(want)
            if ok:
                for g, w in zip(got, want):
                    for gv, wv in zip(g, w):
                        if isinstance(wv, float):
                            if not (gv == wv or abs(gv - wv) <=
                                    1e-9 * max(abs(wv), 1.0)):
                                ok = False
                        elif gv != wv:
                            ok = False
            print(f"q{q}: {'OK' if ok else 'MISMATCH'}", flush=True)
            if not ok:
                bad.append(q)
        print(f"RCCL_W2_RESULT bad={bad}", flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
