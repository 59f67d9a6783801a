"""Multi-process SPMD execution over gloo (CPU, world_size=2).

Verifies the distributed exchange logic (broadcast joins, two-phase
aggregates, gathers) by comparing TPC-H results against single-process
execution on the union of the shards — the engine analogue of the
reference's local-cluster tests (ref: SURVEY §4 tier 3)."""
import json
import os
import pickle
import sys
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

QIDS = list(range(1, 23))


def _worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.exec.context import DistContext

    s = sail_amd.SessionContext(device="cpu")
    s.dist = DistContext(dist, rank=rank, world=world, device="cpu")
    register_tpch(s, sf=0.01, rank=rank, world=world)
    results = {}
    for q in QIDS:
        results[q] = s.sql(QUERIES[q]).collect()
    if rank == 0:
        with open(os.path.join(out_dir, "rank0.pkl"), "wb") as f:
            pickle.dump(results, f)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_tpch_world2_matches_single():
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.engine.executor import concat_columns
    from sail_amd.engine.column import Table

    # single-process truth: union of the two shards
    single = sail_amd.SessionContext(device="cpu")
    shard_tables = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=2).generate_all()
                    for r in range(2)]
    globals_ = None
    for name in shard_tables[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shard_tables[0][name])
            continue
        cols = {}
        for cn in shard_tables[0][name].columns:
            cols[cn] = concat_columns([shard_tables[r][name].columns[cn] for r in range(2)])
        single.catalog.register_table(name, Table(cols))
    want = {q: single.sql(QUERIES[q]).collect() for q in QIDS}

    with tempfile.TemporaryDirectory() as d:
        port = 29512
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker, args=(r, 2, port, d)) for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=540)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        with open(os.path.join(d, "rank0.pkl"), "rb") as f:
            got = pickle.load(f)

    for q in QIDS:
        assert len(got[q]) == len(want[q]), f"q{q}: {len(got[q])} vs {len(want[q])} rows"
        for i, (g, w) in enumerate(zip(got[q], want[q])):
            for gv, wv in zip(g, w):
                if isinstance(wv, float):
                    assert gv == pytest.approx(wv, rel=1e-9, abs=1e-9), f"q{q} row {i}"
                else:
                    assert gv == wv, f"q{q} row {i}: {gv!r} != {wv!r}"


@pytest.mark.timeout(600)
def test_shuffle_join_world2_matches_single():
    """Force the shuffle exchange (threshold=0) and re-check join-heavy
    queries against single-process results."""
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.engine.executor import concat_columns
    from sail_amd.engine.column import Table

    qids = [3, 4, 5, 9, 10, 12, 13, 18, 21]
    single = sail_amd.SessionContext(device="cpu")
    shard_tables = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=2).generate_all()
                    for r in range(2)]
    for name in shard_tables[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shard_tables[0][name])
            continue
        cols = {}
        for cn in shard_tables[0][name].columns:
            cols[cn] = concat_columns([shard_tables[r][name].columns[cn] for r in range(2)])
        single.catalog.register_table(name, Table(cols))
    want = {q: single.sql(QUERIES[q]).collect() for q in qids}

    import tempfile
    with tempfile.TemporaryDirectory() as d:
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker_shuffle, args=(r, 2, 29515, d, qids))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=540)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        with open(os.path.join(d, "rank0.pkl"), "rb") as f:
            got = pickle.load(f)

    for q in qids:
        assert len(got[q]) == len(want[q]), f"q{q} rows"
        for i, (g, w) in enumerate(zip(got[q], want[q])):
            for gv, wv in zip(g, w):
                if isinstance(wv, float):
                    assert gv == pytest.approx(wv, rel=1e-9, abs=1e-9), f"q{q} row {i}"
                else:
                    assert gv == wv, f"q{q} row {i}"


def _worker_shuffle(rank, world, port, out_dir, qids):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.exec.context import DistContext

    s = sail_amd.SessionContext(device="cpu")
    s.dist = DistContext(dist, rank=rank, world=world, device="cpu")
    s.conf["sail.exec.broadcast_threshold_bytes"] = "0"  # force shuffle joins
    s.conf["sail.exec.agg_shuffle_threshold_groups"] = "10"  # force shuffled aggs
    register_tpch(s, sf=0.01, rank=rank, world=world)
    results = {q: s.sql(QUERIES[q]).collect() for q in qids}
    if rank == 0:
        with open(os.path.join(out_dir, "rank0.pkl"), "wb") as f:
            pickle.dump(results, f)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_tpch_world4_subset_matches_single():
    """world=4 (the shape of a half-node): uneven shards, deeper trees.
    Covers rank counts the driver's 4/8-GPU scale runs will use."""
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.engine.executor import concat_columns
    from sail_amd.engine.column import Table

    world = 4
    qids = [1, 3, 5, 9, 18, 21]
    single = sail_amd.SessionContext(device="cpu")
    shard_tables = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=world).generate_all()
                    for r in range(world)]
    for name in shard_tables[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shard_tables[0][name])
            continue
        cols = {}
        for cn in shard_tables[0][name].columns:
            cols[cn] = concat_columns([shard_tables[r][name].columns[cn]
                                       for r in range(world)])
        single.catalog.register_table(name, Table(cols))
    want = {q: single.sql(QUERIES[q]).collect() for q in qids}

    with tempfile.TemporaryDirectory() as d:
        port = 29533
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker_shuffle, args=(r, world, port, d, qids))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=540)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        with open(os.path.join(d, "rank0.pkl"), "rb") as f:
            got = pickle.load(f)

    for q in qids:
        assert len(got[q]) == len(want[q]), f"q{q}: {len(got[q])} vs {len(want[q])}"
        for i, (g, w) in enumerate(zip(got[q], want[q])):
            for gv, wv in zip(g, w):
                if isinstance(wv, float):
                    assert gv == pytest.approx(wv, rel=1e-9, abs=1e-9), f"q{q} row {i}"
                else:
                    assert gv == wv, f"q{q} row {i}: {gv!r} != {wv!r}"


def _worker_scan(rank, world, port, out_dir, qids):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch_parquet
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.exec.context import DistContext

    s = sail_amd.SessionContext(device="cpu")
    s.dist = DistContext(dist, rank=rank, world=world, device="cpu")
    register_tpch_parquet(s, sf=0.01, rank=rank, world=world,
                          data_dir=os.path.join(out_dir, "shards"))
    results = {q: s.sql(QUERIES[q]).collect() for q in qids}
    if rank == 0:
        with open(os.path.join(out_dir, "rank0.pkl"), "wb") as f:
            pickle.dump(results, f)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_tpch_scan_mode_world2_matches_single():
    """Scan-inclusive bench path (parquet shard views, pruned scans) at
    world=2 matches single-process resident execution."""
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.engine.executor import concat_columns
    from sail_amd.engine.column import Table

    qids = [1, 3, 6, 13, 18, 21]
    single = sail_amd.SessionContext(device="cpu")
    shard_tables = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=2).generate_all()
                    for r in range(2)]
    for name in shard_tables[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shard_tables[0][name])
            continue
        cols = {}
        for cn in shard_tables[0][name].columns:
            cols[cn] = concat_columns([shard_tables[r][name].columns[cn] for r in range(2)])
        single.catalog.register_table(name, Table(cols))
    want = {q: single.sql(QUERIES[q]).collect() for q in qids}

    with tempfile.TemporaryDirectory() as d:
        port = 29517
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker_scan, args=(r, 2, port, d, qids))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=540)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        with open(os.path.join(d, "rank0.pkl"), "rb") as f:
            got = pickle.load(f)

    for q in qids:
        assert len(got[q]) == len(want[q]), f"q{q}"
        for i, (g, w) in enumerate(zip(got[q], want[q])):
            for gv, wv in zip(g, w):
                if isinstance(wv, float):
                    assert gv == pytest.approx(wv, rel=1e-9, abs=1e-9), f"q{q} row {i}"
                else:
                    assert gv == wv, f"q{q} row {i}"


def _worker_dist_ops(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    # force the shuffle-based sort/distinct/window paths at tiny scale
    os.environ["SAIL_DIST_SORT_MIN_ROWS"] = "1"
    os.environ["SAIL_DIST_DISTINCT_MIN_ROWS"] = "1"
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch
    from sail_amd.exec.context import DistContext

    s = sail_amd.SessionContext(device="cpu")
    s.dist = DistContext(dist, rank=rank, world=world, device="cpu")
    register_tpch(s, sf=0.01, rank=rank, world=world)
    results = {}
    # range-partitioned distributed sort (multi-key, desc secondary, nulls)
    results["sort"] = s.sql(
        "SELECT l_orderkey, l_linenumber, l_quantity FROM lineitem "
        "ORDER BY l_quantity DESC, l_orderkey, l_linenumber LIMIT 500").collect()
    results["sort_asc"] = s.sql(
        "SELECT o_orderdate, o_orderkey FROM orders "
        "ORDER BY o_orderdate, o_orderkey LIMIT 300").collect()
    # shuffled distinct stays sharded; count over it forces a merge
    results["distinct"] = s.sql(
        "SELECT count(*) FROM (SELECT DISTINCT l_suppkey, l_returnflag "
        "FROM lineitem)").collect()
    results["distinct_rows"] = sorted(s.sql(
        "SELECT DISTINCT l_shipmode, l_returnflag FROM lineitem").collect())
    # partitioned window: each partition lands wholly on one rank
    results["window"] = sorted(s.sql(
        "SELECT o_custkey, o_orderkey, rn FROM ("
        "SELECT o_custkey, o_orderkey, row_number() OVER "
        "(PARTITION BY o_custkey ORDER BY o_orderdate, o_orderkey) AS rn "
        "FROM orders) WHERE rn <= 2 AND o_custkey < 200").collect())
    if rank == 0:
        with open(os.path.join(out_dir, "rank0.pkl"), "wb") as f:
            pickle.dump(results, f)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_shuffled_sort_distinct_window_world2():
    """sort/distinct/window must not rely on whole-table
    gathers — the shuffle-based paths (forced via tiny thresholds) must be
    bit-identical to single-process execution."""
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator
    from sail_amd.engine.executor import concat_columns
    from sail_amd.engine.column import Table

    single = sail_amd.SessionContext(device="cpu")
    shard_tables = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=2).generate_all()
                    for r in range(2)]
    for name in shard_tables[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shard_tables[0][name])
            continue
        cols = {}
        for cn in shard_tables[0][name].columns:
            cols[cn] = concat_columns([shard_tables[r][name].columns[cn] for r in range(2)])
        single.catalog.register_table(name, Table(cols))
    want = {
        "sort": single.sql(
            "SELECT l_orderkey, l_linenumber, l_quantity FROM lineitem "
            "ORDER BY l_quantity DESC, l_orderkey, l_linenumber LIMIT 500").collect(),
        "sort_asc": single.sql(
            "SELECT o_orderdate, o_orderkey FROM orders "
            "ORDER BY o_orderdate, o_orderkey LIMIT 300").collect(),
        "distinct": single.sql(
            "SELECT count(*) FROM (SELECT DISTINCT l_suppkey, l_returnflag "
            "FROM lineitem)").collect(),
        "distinct_rows": sorted(single.sql(
            "SELECT DISTINCT l_shipmode, l_returnflag FROM lineitem").collect()),
        "window": sorted(single.sql(
            "SELECT o_custkey, o_orderkey, rn FROM ("
            "SELECT o_custkey, o_orderkey, row_number() OVER "
            "(PARTITION BY o_custkey ORDER BY o_orderdate, o_orderkey) AS rn "
            "FROM orders) WHERE rn <= 2 AND o_custkey < 200").collect()),
    }
    with tempfile.TemporaryDirectory() as d:
        port = 29519
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker_dist_ops, args=(r, 2, port, d))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=540)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        with open(os.path.join(d, "rank0.pkl"), "rb") as f:
            got = pickle.load(f)
    for k in want:
        assert got[k] == want[k], k


def test_fake_rccl_dist_world2_cpu():
    """The thread-rank fake communicator (used to validate device
    collectives on a 1-GPU box) matches single-process results on CPU."""
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator, register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.engine.column import Table
    from sail_amd.engine.executor import concat_columns
    from sail_amd.exec.context import DistContext
    from sail_amd.exec.fake_dist import run_world

    world = 2
    qids = [1, 6, 13]
    results = {}

    def body(rank, dist):
        s = sail_amd.SessionContext(device="cpu")
        s.dist = DistContext(dist, rank=rank, world=world, device="cpu")
        register_tpch(s, sf=0.01, rank=rank, world=world)
        results[rank] = {q: s.sql(QUERIES[q]).collect() for q in qids}

    run_world(world, "cpu", body, strict_cuda=False)

    single = sail_amd.SessionContext(device="cpu")
    shards = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=world).generate_all()
              for r in range(world)]
    for name in shards[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shards[0][name])
            continue
        cols = {cn: concat_columns([shards[r][name].columns[cn]
                                    for r in range(world)])
                for cn in shards[0][name].columns}
        single.catalog.register_table(name, Table(cols))
    for q in qids:
        want = single.sql(QUERIES[q]).collect()
        assert results[0][q] == want and results[1][q] == want, q
