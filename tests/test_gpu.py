"""GPU tests (MI355X): kernel numerics vs CPU reference + E2E TPC-H parity.

Every test here compares the HIP path against the torch/host reference on
identical inputs.
"""
import random
import re

import pytest
import torch

import sail_amd
from sail_amd.engine import types as T
from sail_amd.engine.column import StringColumn

pytestmark = pytest.mark.gpu


def _rand_strings(n, words=("special", "requests", "foo", "ba%r", "x_y", ""), seed=7):
    rng = random.Random(seed)
    out = []
    for _ in range(n):
        k = rng.randint(0, 6)
        out.append(" ".join(rng.choice(words) for _ in range(k)))
    return out


@pytest.fixture(scope="module")
def ext():
    from sail_amd.ops import kernels

    return kernels.require()


def test_like_mask_matches_cpu(ext):
    vals = _rand_strings(5000)
    cpu = StringColumn.from_pylist(vals, device="cpu", dict_encode=False)
    gpu = cpu.to("cuda")
    from sail_amd.engine.eval import like_to_regex

    for pat in ["%special%requests%", "special%", "%x_y", "%foo%", "sp_cial%", "%", ""]:
        rx = like_to_regex(pat)
        want = [rx.match(v) is not None for v in vals]
        got = ext.like_mask(gpu.offsets, gpu.bytes_, pat.encode()).cpu().tolist()
        assert got == want, pat


def test_string_hash_matches_cpu(ext):
    from sail_amd.engine.joins import fnv_key_tensor

    vals = _rand_strings(3000, words=("alpha", "beta", "gamma", "delta-longer-word"))
    cpu = StringColumn.from_pylist(vals, device="cpu", dict_encode=False)
    gpu = cpu.to("cuda")
    want = fnv_key_tensor(cpu)
    got = ext.string_hash64(gpu.offsets, gpu.bytes_).cpu()
    assert torch.equal(want, got)


def test_substr_fixed(ext):
    vals = ["abcdef", "x", "", "hello world"]
    cpu = StringColumn.from_pylist(vals, device="cpu", dict_encode=False)
    gpu = cpu.to("cuda")
    buf, lens = ext.substr_fixed(gpu.offsets, gpu.bytes_, 1, 3)
    assert lens.cpu().tolist() == [3, 0, 0, 3]
    assert bytes(buf.cpu().numpy().tobytes()[:3]) == b"bcd"


def test_sql_basics_on_gpu():
    s = sail_amd.SessionContext(device="cuda")
    s.create_dataframe({"a": [1, 2, 3, 4], "b": [1.0, 2.0, 3.0, 4.0],
                        "c": ["x", "y", "x", "z"]}, name="t")
    assert s.sql("SELECT c, sum(a) FROM t GROUP BY c ORDER BY c").collect() == [
        ("x", 4), ("y", 2), ("z", 4)]
    assert s.sql("SELECT a FROM t WHERE b > 2.5 ORDER BY a DESC").collect() == [(4,), (3,)]


@pytest.fixture(scope="module")
def tpch_pair():
    """Same data on CPU and GPU (generated on CPU, copied) for result parity."""
    from sail_amd.datagen.tpch import TpchGenerator

    cpu = sail_amd.SessionContext(device="cpu")
    gpu = sail_amd.SessionContext(device="cuda")
    gen = TpchGenerator(sf=0.01, device="cpu")
    tables = gen.generate_all()
    for name, tbl in tables.items():
        cpu.catalog.register_table(name, tbl)
        gpu.catalog.register_table(name, tbl.to("cuda"))
    return cpu, gpu


@pytest.mark.parametrize("q", list(range(1, 23)))
def test_tpch_gpu_matches_cpu(tpch_pair, q):
    from sail_amd.datagen.tpch_queries import QUERIES

    cpu, gpu = tpch_pair
    want = cpu.sql(QUERIES[q]).collect()
    got = gpu.sql(QUERIES[q]).collect()
    assert len(got) == len(want), f"q{q} row count"
    for i, (g, w) in enumerate(zip(got, want)):
        for gv, wv in zip(g, w):
            if isinstance(wv, float):
                assert gv == pytest.approx(wv, rel=1e-9, abs=1e-9), f"q{q} row {i}"
            else:
                assert gv == wv, f"q{q} row {i}"


def test_grouped_acc_kernel_matches_torch(ext):
    torch.manual_seed(0)
    n = 1_000_000
    for G in (4, 7, 64, 1500):
        gid = torch.randint(0, G, (n,), dtype=torch.int32, device="cuda")
        vi = torch.randint(-1000, 1000, (n,), dtype=torch.int64, device="cuda")
        vf = torch.rand(n, dtype=torch.float64, device="cuda")
        mask = (torch.rand(n, device="cuda") < 0.7)
        out = ext.grouped_acc(gid, mask, [vi, vf, None], [0, 1, 2], G)
        g64 = gid.to(torch.int64)
        m = mask
        want_i = torch.zeros(G, dtype=torch.int64, device="cuda")
        want_i.index_add_(0, g64[m], vi[m])
        want_f = torch.zeros(G, dtype=torch.float64, device="cuda")
        want_f.index_add_(0, g64[m], vf[m])
        want_c = torch.zeros(G, dtype=torch.int64, device="cuda")
        want_c.index_add_(0, g64[m], torch.ones(int(m.sum()), dtype=torch.int64, device="cuda"))
        assert torch.equal(out[0], want_i), G
        assert torch.allclose(out[1].view(torch.float64), want_f, rtol=1e-12), G
        assert torch.equal(out[2], want_c), G
        # min/max via lds variant
        out2 = ext.grouped_acc(gid, None, [vi, vi], [3, 4], G)
        wmin = torch.full((G,), 2**62, dtype=torch.int64, device="cuda")
        wmin.scatter_reduce_(0, g64, vi, reduce="amin", include_self=True)
        wmax = torch.full((G,), -2**62, dtype=torch.int64, device="cuda")
        wmax.scatter_reduce_(0, g64, vi, reduce="amax", include_self=True)
        assert torch.equal(out2[0], wmin), G
        assert torch.equal(out2[1], wmax), G


def test_fused_agg_end_to_end_gpu():
    import sail_amd
    from sail_amd.engine import types as T

    s = sail_amd.SessionContext(device="cuda")
    import random

    rng = random.Random(3)
    n = 200_000
    ks = [rng.randint(0, 5) for _ in range(n)]
    vs = [rng.randint(-100, 100) for _ in range(n)]
    ds = [round(rng.uniform(0, 100), 2) for _ in range(n)]
    s.create_dataframe({"k": ks, "v": vs, "d": ds},
                       schema={"k": T.I32, "v": T.I64, "d": T.DecimalType(12, 2)}, name="t")
    got = s.sql("SELECT k, sum(v), count(*), avg(d), min(v), max(v) FROM t GROUP BY k ORDER BY k").collect()
    import collections

    acc = collections.defaultdict(lambda: [0, 0, 0.0, 10**9, -10**9])
    for k, v, d in zip(ks, vs, ds):
        a = acc[k]
        a[0] += v
        a[1] += 1
        a[2] += d
        a[3] = min(a[3], v)
        a[4] = max(a[4], v)
    for row in got:
        k, sv, cnt, avgd, mn, mx = row
        a = acc[k]
        assert sv == a[0] and cnt == a[1] and mn == a[3] and mx == a[4]
        assert abs(avgd - a[2] / a[1]) < 1e-4


def test_hash_join_kernels_match_sort_join(ext):
    torch.manual_seed(1)
    for nb, np_, kr in ((1000, 5000, 300), (100000, 400000, 50000), (7, 3, 4)):
        bids = torch.randint(0, kr, (nb,), dtype=torch.int64, device="cuda")
        pids = torch.randint(0, kr, (np_,), dtype=torch.int64, device="cuda")
        from sail_amd.engine.joins import _expand_matches_gpu

        p_idx, b_idx, counts = _expand_matches_gpu(bids, pids, ext)
        # reference: sort-based matcher on CPU
        bc, pc = bids.cpu(), pids.cpu()
        order = torch.argsort(bc)
        bs = bc[order]
        lo = torch.searchsorted(bs, pc, right=False)
        hi = torch.searchsorted(bs, pc, right=True)
        want_counts = (hi - lo)
        assert torch.equal(counts.cpu(), want_counts)
        # pair set equality (order within a key is arbitrary)
        got = set(zip(p_idx.cpu().tolist(), b_idx.cpu().tolist()))
        want = set()
        for i in range(np_):
            for j in range(int(lo[i]), int(hi[i])):
                want.add((i, int(order[j])))
        assert got == want, (nb, np_, kr)


def test_merge_into_on_gpu(tmp_path):
    import sail_amd
    from sail_amd.engine import types as T

    s = sail_amd.SessionContext(device="cuda")
    s.create_dataframe({"id": list(range(1000)), "v": [float(i) for i in range(1000)]},
                       schema={"id": T.I64, "v": T.F64}, name="seed")
    p = str(tmp_path / "dt")
    s.table("seed").write.format("delta").mode("overwrite").save(p)
    s.create_dataframe({"id": [5, 2000], "v": [555.0, -1.0]},
                       schema={"id": T.I64, "v": T.F64}, name="src")
    s.sql(f"MERGE INTO delta.`{p}` t USING src u ON t.id = u.id "
          "WHEN MATCHED THEN UPDATE SET v = u.v "
          "WHEN NOT MATCHED THEN INSERT (id, v) VALUES (u.id, u.v)")
    rows = dict(s.read.format("delta").load(p).collect())
    assert rows[5] == 555.0 and rows[2000] == -1.0 and len(rows) == 1001


def test_update_delete_on_gpu():
    import sail_amd
    from sail_amd.engine import types as T

    s = sail_amd.SessionContext(device="cuda")
    s.create_dataframe({"id": [1, 2, 3], "v": [1.0, 2.0, 3.0]},
                       schema={"id": T.I64, "v": T.F64}, name="t")
    s.sql("UPDATE t SET v = v * 10 WHERE id > 1")
    s.sql("DELETE FROM t WHERE v >= 30.0")
    assert sorted(s.sql("SELECT * FROM t").collect()) == [(1, 1.0), (2, 20.0)]


@pytest.mark.gpu
def test_array_functions_on_gpu():
    """Segment ops (gather/cumsum/scatter_reduce) over ListColumn on device."""
    import sail_amd

    s = sail_amd.SessionContext(device="cuda")
    s.create_dataframe({"k": ["a", "b"], "v": [3, 5]}, name="t")
    rows = s.sql("SELECT k, explode(sequence(1, v)) AS e FROM t ORDER BY k, e").collect()
    assert rows == [("a", 1), ("a", 2), ("a", 3),
                    ("b", 1), ("b", 2), ("b", 3), ("b", 4), ("b", 5)]
    rows = s.sql("SELECT k, collect_list(x) FROM "
                 "(SELECT k, explode(sequence(1, v)) AS x FROM t) e "
                 "GROUP BY k ORDER BY k").collect()
    assert rows == [("a", [1, 2, 3]), ("b", [1, 2, 3, 4, 5])]
    rows = s.sql("SELECT array_min(sequence(v, 1)), array_position(sequence(1, v), 4), "
                 "sort_array(array(3, 1, 2)) FROM t WHERE k = 'b'").collect()
    assert rows == [(1, 4, [1, 2, 3])]


@pytest.mark.gpu
def test_streaming_incremental_agg_on_gpu():
    """Micro-batch streaming with device-resident aggregation state."""
    import sail_amd
    from sail_amd.engine import types as T

    s = sail_amd.SessionContext(device="cuda")
    sdf = s.read_stream.format("memory").schema(
        {"k": T.STRING, "v": T.I64}).load(name="ev")
    src = sdf.source
    q = (sdf.sql("SELECT k, sum(v) AS sv, count(*) AS n FROM ev GROUP BY k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("agg_gpu").trigger(processing_time=0.01).start())
    assert q._mode == "incremental"
    src.add_rows({"k": ["a", "b", "a"], "v": [1, 2, 3]})
    q.process_all_available()
    src.add_rows({"k": ["b", "c"], "v": [10, 5]})
    q.process_all_available()
    q.stop()
    assert q.exception is None
    rows = dict((r[0], (r[1], r[2]))
                for r in s.sql("SELECT k, sv, n FROM agg_gpu").collect())
    assert rows == {"a": (4, 2), "b": (12, 2), "c": (5, 1)}


@pytest.mark.gpu
def test_tablesample_gpu_deterministic():
    import sail_amd

    s = sail_amd.SessionContext(device="cuda")
    a = s.sql("SELECT sum(id) FROM range(100000) TABLESAMPLE (10 PERCENT) REPEATABLE (3) t").collect()
    b = s.sql("SELECT sum(id) FROM range(100000) TABLESAMPLE (10 PERCENT) REPEATABLE (3) t").collect()
    assert a == b and a[0][0] > 0


@pytest.mark.gpu
def test_direct_aggregate_gpu_parity():
    """q18-shaped dense high-cardinality aggregate: direct-address path vs
    generic gid path on device."""
    import sail_amd
    import sail_amd.engine.aggregates as agg_mod
    import torch

    s = sail_amd.SessionContext(device="cuda")
    n = 5_000_000
    g = torch.Generator().manual_seed(9)
    keys = torch.randint(0, 1_000_000, (n,), generator=g)
    vals = torch.randint(-50, 50, (n,), generator=g)
    from sail_amd.engine.column import Column, Table
    from sail_amd.engine import types as T

    s.catalog.register_table("dk", Table({
        "k": Column(T.I64, keys.cuda()), "v": Column(T.I64, vals.cuda())}))
    q = "SELECT k, sum(v), count(*) FROM dk GROUP BY k ORDER BY k LIMIT 50"
    old = agg_mod.DIRECT_MIN_ROWS
    try:
        agg_mod.DIRECT_MIN_ROWS = 1 << 60
        want = s.sql(q).collect()
        agg_mod.DIRECT_MIN_ROWS = 1
        got = s.sql(q).collect()
    finally:
        agg_mod.DIRECT_MIN_ROWS = old
    assert got == want


@pytest.mark.gpu
def test_hash_group_ids_matches_sort_path():
    """hg_group kernel vs the unique/searchsorted path: same partitioning
    (group numbering may differ — compare per-group aggregate results)."""
    import sail_amd
    import torch
    from sail_amd.engine.column import Column, Table
    from sail_amd.engine import types as T

    s = sail_amd.SessionContext(device="cuda")
    n = 3_000_000
    g = torch.Generator().manual_seed(11)
    # sparse domain: huge random int64 keys force the hash path
    keys = torch.randint(-(1 << 60), 1 << 60, (n,), generator=g)
    vals = torch.randint(0, 1000, (n,), generator=g)
    # inject duplicates so groups have >1 row
    keys[n // 2:] = keys[: n - n // 2]
    s.catalog.register_table("hg", Table({
        "k": Column(T.I64, keys.cuda()), "v": Column(T.I64, vals.cuda())}))
    rows = s.sql("SELECT k, sum(v) AS sv, count(*) AS c FROM hg GROUP BY k "
                 "ORDER BY sv DESC, k LIMIT 100").collect()
    # CPU truth
    s2 = sail_amd.SessionContext(device="cpu")
    s2.catalog.register_table("hg", Table({
        "k": Column(T.I64, keys), "v": Column(T.I64, vals)}))
    want = s2.sql("SELECT k, sum(v) AS sv, count(*) AS c FROM hg GROUP BY k "
                  "ORDER BY sv DESC, k LIMIT 100").collect()
    assert rows == want


@pytest.mark.gpu
def test_feature_batch_on_gpu():
    """Device coverage for the round's feature additions in one session:
    array set ops, map lambdas, window frames, topk, pivot, recursive CTE."""
    import sail_amd

    s = sail_amd.SessionContext(device="cuda")
    assert s.sql("SELECT array_union(array(1,2), array(2,3)), "
                 "array_except(array(1,2), array(2,3))").collect() == [([1, 2, 3], [1])]
    assert s.sql("SELECT transform(sequence(1, 4), x -> x * x)").collect() == [([1, 4, 9, 16],)]
    assert s.sql("SELECT map_filter(map('a',1,'b',2), (k,v) -> v > 1)").collect() == [({"b": 2},)]
    s.create_dataframe({"o": [1, 2, 3, 4], "v": [10, 20, 30, 40]}, name="wf")
    assert s.sql("SELECT o, sum(v) OVER (ORDER BY o ROWS BETWEEN 1 PRECEDING AND "
                 "1 FOLLOWING) FROM wf ORDER BY o").collect() == [
        (1, 30), (2, 60), (3, 90), (4, 70)]
    rows = s.sql("WITH RECURSIVE n(x) AS (SELECT 1 UNION ALL SELECT x + 1 FROM n "
                 "WHERE x < 100) SELECT sum(x) FROM n").collect()
    assert rows == [(5050,)]
    s.create_dataframe({"g": ["x", "x", "y"], "k": ["a", "b", "a"], "v": [1, 2, 3]},
                       name="pv")
    assert s.sql("SELECT * FROM pv PIVOT (sum(v) FOR k IN ('a', 'b')) ORDER BY g").collect() == [
        ("x", 1, 2), ("y", 3, None)]
    # topk path over a big range (selection, not full sort)
    assert s.sql("SELECT id FROM range(5000000) ORDER BY id DESC LIMIT 3").collect() == [
        (4999999,), (4999998,), (4999997,)]
    # struct/json
    assert s.sql("SELECT to_json(named_struct('a', 1))").collect() == [('{"a":1}',)]


# -- GPU parquet decode (datasource/gpu_parquet.py + parquet_decode.hip) ----

def test_gpu_parquet_decode_matches_pyarrow(tmp_path):
    """End-to-end device decode vs pyarrow on a file exercising every
    supported encoding (DELTA_BINARY_PACKED, DELTA_LENGTH_BYTE_ARRAY,
    RLE_DICTIONARY, PLAIN, FLBA decimal, nulls, multiple row groups)."""
    from test_gpu_parquet import _write_mixed
    from sail_amd.datasource import gpu_parquet as G

    for nulls in (False, True):
        p = str(tmp_path / f"mix{int(nulls)}.parquet")
        t = _write_mixed(p, nrows=50_000, page_size=4096, nulls=nulls,
                         row_groups=3)
        out = G.read_gpu([p], [(f.name, None) for f in t.schema], "cuda:0")
        torch.cuda.synchronize()
        for name in t.schema.names:
            col = out.columns[name]
            assert str(col.device).startswith("cuda")
            got = col.to_pylist()
            exp = t.column(name).to_pylist()
            if name == "d":
                exp = [None if v is None else float(v) for v in exp]
                assert got == pytest.approx(exp), name
            elif name == "f":
                assert got == pytest.approx(exp), name
            elif name == "dt":
                assert [str(x) for x in got] == [str(x) for x in exp], name
            else:
                assert got == exp, name


def test_gpu_parquet_plain_bytearray(tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq
    from sail_amd.datasource import gpu_parquet as G

    vals = [f"value-{i}-{'y' * (i % 29)}" for i in range(30_000)]
    p = str(tmp_path / "pb.parquet")
    pq.write_table(pa.table({"s": pa.array(vals)}), p, compression="NONE",
                   use_dictionary=False, data_page_size=8192,
                   data_page_version="1.0")
    out = G.read_gpu([p], [("s", None)], "cuda:0")
    assert out.columns["s"].to_pylist() == vals


def test_gpu_parquet_sql_roundtrip(tmp_path, gpu_session):
    """parquet.`path` SQL scans decode on device (force mode: no host
    fallback allowed)."""
    import os
    import pyarrow as pa
    import pyarrow.parquet as pq

    s = gpu_session
    p = str(tmp_path / "t.parquet")
    n = 100_000
    pq.write_table(
        pa.table({"k": pa.array(range(n), pa.int64()),
                  "g": pa.array(["a", "b", "c", "d"][0:1] * n),
                  "v": pa.array([float(i % 97) for i in range(n)])}),
        p, compression="NONE", use_dictionary=["g"],
        column_encoding={"k": "DELTA_BINARY_PACKED"},
        data_page_version="1.0")
    os.environ["SAIL_IO_GPU_PARQUET"] = "force"
    try:
        r = s.sql(f"SELECT g, count(*), sum(v), max(k) FROM parquet.`{p}` "
                  "GROUP BY g").collect()
        assert r == [("a", n, sum(float(i % 97) for i in range(n)), n - 1)]
    finally:
        os.environ["SAIL_IO_GPU_PARQUET"] = "auto"


def test_gpu_parquet_dict_lex_sorted(tmp_path):
    """Scanned dictionaries must come back lex-sorted (engine invariant:
    dict code order == byte order, relied on by min/max/compare/sort) —
    parquet dictionary pages are first-occurrence ordered on disk."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from sail_amd.datasource import gpu_parquet as G

    p = str(tmp_path / "dl.parquet")
    vals = [f"w{(i * 131) % 997:03d}" for i in range(60_000)]
    pq.write_table(pa.table({"s": pa.array(vals)}), p, compression="NONE",
                   use_dictionary=["s"], data_page_version="1.0",
                   row_group_size=20_000)
    for _ in range(2):  # second read exercises the device dict cache
        out = G.read_gpu([p], [("s", None)], "cuda:0")
        c = out.columns["s"]
        assert c.is_dict
        dv = c.dict_values()
        assert dv == sorted(dv)
        assert c.to_pylist() == vals


def test_string_hash2_and_pairs_equal_match_cpu(ext):
    from sail_amd.engine import joins

    vals = [f"k-{i%97}-{'abc'*(i%11)}" for i in range(20000)]
    cpu = StringColumn.from_pylist(vals, device="cpu", dict_encode=False)
    gpu = cpu.to("cuda")
    assert joins.hash2_tensor(gpu).cpu().tolist() == \
        joins.hash2_tensor(cpu).tolist()
    ia = torch.arange(0, 20000, 2)
    ib = torch.arange(1, 20000, 2)
    g = joins.str_pairs_equal(gpu, ia.cuda(), gpu, ib.cuda()).cpu()
    c = joins.str_pairs_equal(cpu, ia, cpu, ib)
    assert torch.equal(g, c)


def test_exact_string_codes_gpu_matches_cpu(ext):
    from sail_amd.engine import joins

    vals = [f"value-{i % 513}-{'pad' * (i % 7)}" for i in range(100_000)]
    cpu = StringColumn.from_pylist(vals, device="cpu", dict_encode=False)
    gpu = cpu.to("cuda")
    cc = joins.exact_string_codes([cpu])[0].tolist()
    gc = joins.exact_string_codes([gpu])[0].cpu().tolist()
    # code NUMBERS may differ (sort ties); the partition must be identical
    def parts(codes):
        m = {}
        for i, c in enumerate(codes):
            m.setdefault(c, []).append(i)
        return sorted(map(tuple, m.values()))
    assert parts(cc) == parts(gc)
    assert len(set(gc)) == len(set(vals))


def test_gpu_parquet_large_pages_batched_runs(tmp_path):
    """Pages with >>kRuns RLE runs and >>kRuns delta miniblocks exercise the
    kernels' LDS batching loops (the SF100 barrier-race regression)."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from sail_amd.datasource import gpu_parquet as G

    n = 2_000_000
    rng = np.random.default_rng(3)
    ints = rng.integers(-10**14, 10**14, n)
    # alternating short runs -> tens of thousands of RLE runs per page
    dict_vals = np.array(["aaaa", "bbbb", "cccc"])[(np.arange(n) // 9) % 3]
    p = str(tmp_path / "big.parquet")
    pq.write_table(
        pa.table({"k": pa.array(ints), "g": pa.array(dict_vals),
                  "ks": pa.array(np.sort(rng.integers(0, 10**10, n)))}),
        p, compression="NONE", use_dictionary=["g"],
        column_encoding={"k": "DELTA_BINARY_PACKED",
                         "ks": "DELTA_BINARY_PACKED"},
        data_page_size=1 << 20, data_page_version="1.0",
        row_group_size=n)
    out = G.read_gpu([p], [("k", None), ("g", None), ("ks", None)], "cuda:0")
    torch.cuda.synchronize()
    assert out.columns["k"].data.cpu().numpy().tolist() == ints.tolist()
    assert out.columns["ks"].data.cpu().numpy().min() >= 0
    import numpy as _np
    assert _np.array_equal(out.columns["ks"].data.cpu().numpy(),
                           _np.sort(rng.integers(0, 10**10, 0)) if False else
                           pq.read_table(p, columns=["ks"]).column("ks").to_numpy())
    got_g = out.columns["g"]
    assert got_g.is_dict
    assert got_g.to_pylist() == dict_vals.tolist()


def test_spmd_exchanges_world2_on_one_gpu():
    """The RCCL exchange paths (hash shuffle, range sort, gathers, shuffled
    distinct/window) executed with DEVICE tensors: RCCL refuses 2 ranks on
    one GPU, so a 2-thread fake communicator with RCCL's preconditions
    (cuda+contiguous tensors, exact split sums, dtype matches) drives the
    backend=="nccl" branches. Results must match single-GPU execution."""
    import os

    os.environ["SAIL_DIST_SORT_MIN_ROWS"] = "1"
    os.environ["SAIL_DIST_DISTINCT_MIN_ROWS"] = "1"

    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator, register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.engine.column import Table
    from sail_amd.engine.executor import Executor, concat_columns
    from sail_amd.exec.context import DistContext
    from sail_amd.exec.fake_dist import run_world

    # force the shuffle paths through env-read class attrs
    Executor.DIST_SORT_MIN_ROWS = 1
    Executor.DIST_DISTINCT_MIN_ROWS = 1

    world = 2
    qids = [1, 3, 5, 6, 13, 18, 21]
    results = {}

    def body(rank, dist):
        s = sail_amd.SessionContext(device="cuda:0")
        s.conf["sail.exec.broadcast_threshold_bytes"] = "65536"
        s.conf["sail.exec.agg_shuffle_threshold_groups"] = "64"
        s.dist = DistContext(dist, rank=rank, world=world, device="cuda:0")
        register_tpch(s, sf=0.05, rank=rank, world=world)
        out = {}
        for q in qids:
            out[q] = s.sql(QUERIES[q]).collect()
        results[rank] = out

    run_world(world, "cuda:0", body)

    single = sail_amd.SessionContext(device="cuda:0")
    shards = [TpchGenerator(sf=0.05, device="cuda:0", rank=r,
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
    for q in qids:
        want = single.sql(QUERIES[q]).collect()
        for r in range(world):
            got = results[r][q]
            assert len(got) == len(want), f"q{q} rank{r}"
            for g, w in zip(got, want):
                for gv, wv in zip(g, w):
                    if isinstance(wv, float):
                        assert gv == pytest.approx(wv, rel=1e-9, abs=1e-9), \
                            f"q{q} rank{r}"
                    else:
                        assert gv == wv, f"q{q} rank{r}: {gv!r} != {wv!r}"


def test_mfma_sum_f64_measurement(ext):
    """MFMA-assisted reduction (north star: 'MFMA for aggregate
    reductions'): correctness vs torch.sum plus an A/B timing against the
    VALU kernel — the result is recorded in profiles/README.md whichever
    way it lands (a whole-column sum is HBM-bound, so parity is the
    expected outcome)."""
    import time

    n = 200_000_000
    x = torch.rand(n, dtype=torch.float64, device="cuda")
    want = float(x.sum().item())
    got_mfma = float(ext.mfma_sum_f64(x, True).item())
    got_valu = float(ext.mfma_sum_f64(x, False).item())
    assert got_mfma == pytest.approx(want, rel=1e-9)
    assert got_valu == pytest.approx(want, rel=1e-9)

    def bench(fn, reps=10):
        fn()  # warm
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.time() - t0) / reps

    t_mfma = bench(lambda: ext.mfma_sum_f64(x, True))
    t_valu = bench(lambda: ext.mfma_sum_f64(x, False))
    t_torch = bench(lambda: x.sum())
    gbps = n * 8 / 1e9
    print(f"\n# mfma_sum_f64 A/B on {n} doubles ({gbps:.1f} GB): "
          f"mfma={t_mfma*1e3:.2f}ms ({gbps/t_mfma:.0f} GB/s) "
          f"valu={t_valu*1e3:.2f}ms ({gbps/t_valu:.0f} GB/s) "
          f"torch={t_torch*1e3:.2f}ms ({gbps/t_torch:.0f} GB/s)")
    # both must run at a credible fraction of HBM bandwidth
    assert gbps / t_mfma > 1000 and gbps / t_valu > 1000


def test_gpu_streamed_scan_and_join_aggregate(tmp_path, gpu_session,
                                              monkeypatch):
    """Out-of-core paths on device: scan-agg and scan-join-agg stream
    row-group batches through GPU decode with a bounded state."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from sail_amd.engine.executor import Executor

    n = 1_000_000
    rng = np.random.default_rng(11)
    fact = str(tmp_path / "fact.parquet")
    pq.write_table(pa.table({
        "k": pa.array((np.arange(n) % 101).astype("int64")),
        "v": pa.array(rng.integers(0, 1000, n)),
    }), fact, compression="NONE", row_group_size=100_000,
        data_page_version="1.0")
    dim = str(tmp_path / "dim.parquet")
    pq.write_table(pa.table({
        "k": pa.array(list(range(101)), pa.int64()),
        "grp": pa.array(["g" + str(i % 5) for i in range(101)]),
    }), dim, compression="NONE", data_page_version="1.0")

    s = gpu_session
    agg_sql = (f"SELECT k % 10 m, count(*) c, sum(v) sv FROM "
               f"parquet.`{fact}` GROUP BY k % 10 ORDER BY m")
    join_sql = (f"SELECT d.grp, count(*) c, sum(f.v) sv FROM "
                f"parquet.`{fact}` f JOIN parquet.`{dim}` d ON f.k = d.k "
                f"GROUP BY d.grp ORDER BY d.grp")
    want_agg = s.sql(agg_sql).collect()
    want_join = s.sql(join_sql).collect()
    monkeypatch.setattr(Executor, "STREAM_SCAN_BYTES", 1_000_000)
    monkeypatch.setattr(Executor, "STREAM_SCAN_BATCH_ROWS", 200_000)
    assert s.sql(agg_sql).collect() == want_agg
    assert s.sql(join_sql).collect() == want_join


def test_decode_determinism_stress(tmp_path):
    """Race detector for the batched decode kernels (the
    MI355X analogue of sanitizer jobs). The round-2 barrier race in the
    batched-page loops produced NONdeterministic corruption under
    scheduler variation; decoding the same file repeatedly with varying
    batch shapes must be bit-identical."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from sail_amd.datasource import gpu_parquet as G

    n = 400_000
    rng = np.random.default_rng(5)
    p = str(tmp_path / "det.parquet")
    pq.write_table(pa.table({
        "a": pa.array(rng.integers(-10**9, 10**9, n)),
        "s": pa.array([f"w{int(x)%997:03d}" for x in rng.integers(0, 10**6, n)]),
        "r": pa.array([f"raw-{int(x)}" for x in rng.integers(0, 10**6, n)]),
    }), p, compression="NONE", row_group_size=50_000,
        use_dictionary=["s"], data_page_version="1.0",
        column_encoding={"a": "DELTA_BINARY_PACKED",
                         "r": "DELTA_LENGTH_BYTE_ARRAY"})
    schema = [("a", None), ("s", None), ("r", None)]

    def snap():
        out = G.read_gpu([p], schema, "cuda:0")
        torch.cuda.synchronize()
        return {k: (c.to_pylist()) for k, c in out.columns.items()}

    base = snap()
    for _ in range(4):
        G._INDEX_CACHE.clear()  # force fresh page tables + dict decode
        G._DICT_CACHE.clear()
        again = snap()
        for k in base:
            assert again[k] == base[k], f"nondeterministic decode: {k}"
