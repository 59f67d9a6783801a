"""GPU parquet decode path: page-index parsing (CPU), host orchestration
against the numpy kernel simulator (CPU), and end-to-end correctness
vs pyarrow (GPU, tests/test_gpu.py has the device half)."""
import decimal
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from sail_amd.datasource import gpu_parquet as G

import pq_sim


def _write_mixed(path, nrows=5000, page_size=700, nulls=False, compression="NONE",
                 row_groups=1):
    rng = np.random.default_rng(7)
    ints = rng.integers(-10**12, 10**12, nrows)
    sorted_ints = np.sort(rng.integers(0, 10**9, nrows))
    decs = [decimal.Decimal(int(v)) / 100 for v in rng.integers(-10**6, 10**6, nrows)]
    dates = rng.integers(8000, 12000, nrows).astype("int32")
    lows = np.array(["AIR", "FOB", "MAIL", "RAIL", "SHIP"])[
        rng.integers(0, 5, nrows)]
    his = np.array([f"s{v}_" + "x" * int(v % 23) for v in rng.integers(0, 10**9, nrows)])
    floats = rng.standard_normal(nrows)
    i32 = rng.integers(-2**31, 2**31 - 1, nrows).astype("int32")

    def _mask(arr):
        if not nulls:
            return arr
        m = rng.random(nrows) < 0.1
        return pa.array(arr, mask=m)

    t = pa.table({
        "k": _mask(ints),
        "ks": pa.array(sorted_ints, pa.int64()),
        "d": pa.array(decs, pa.decimal128(12, 2)),
        "dt": pa.array(dates, pa.date32()),
        "s_low": _mask(lows),
        "s_hi": _mask(his),
        "f": pa.array(floats, pa.float64()),
        "i32": pa.array(i32, pa.int32()),
    })
    pq.write_table(
        t, path, compression=compression,
        use_dictionary=["s_low"],
        column_encoding={"k": "DELTA_BINARY_PACKED",
                         "ks": "DELTA_BINARY_PACKED",
                         "s_hi": "DELTA_LENGTH_BYTE_ARRAY"},
        data_page_size=page_size, data_page_version="1.0",
        row_group_size=max(nrows // row_groups, 1))
    return t


def test_file_index_matches_pyarrow(tmp_path):
    p = str(tmp_path / "m.parquet")
    _write_mixed(p, nrows=4000, row_groups=3)
    idx = G.file_index(p)
    md = pq.ParquetFile(p).metadata
    assert idx.num_rows == md.num_rows
    for name in ("k", "d", "s_low", "s_hi"):
        ci = idx.column_index(name)
        chunks = idx.chunks(ci)
        assert len(chunks) == md.num_row_groups
        for rg, ch in enumerate(chunks):
            cmd = md.row_group(rg).column(ci)
            assert sum(pg.nvals for pg in ch.pages) == cmd.num_values
            lo = (cmd.dictionary_page_offset
                  if cmd.dictionary_page_offset is not None
                  else cmd.data_page_offset)
            assert ch.start == lo
            assert ch.end == lo + cmd.total_compressed_size
    # index is cached
    assert G.file_index(p) is idx


def test_compressed_file_unsupported(tmp_path):
    p = str(tmp_path / "sn.parquet")
    _write_mixed(p, compression="snappy")
    idx = G.file_index(p)
    with pytest.raises(G.Unsupported):
        idx.chunks(0)


@pytest.fixture()
def sim(monkeypatch):
    monkeypatch.setattr(G, "_ext", lambda: pq_sim)
    monkeypatch.setattr(G, "_ALLOW_CPU", True)
    yield


def _expect_col(t: pa.Table, name: str):
    col = t.column(name)
    return col.to_pylist()


@pytest.mark.parametrize("nulls", [False, True])
@pytest.mark.parametrize("row_groups", [1, 3])
def test_sim_decode_matches_pyarrow(tmp_path, sim, nulls, row_groups):
    p = str(tmp_path / "sim.parquet")
    t = _write_mixed(p, nrows=3000, page_size=600, nulls=nulls,
                     row_groups=row_groups)
    schema = [(f.name, None) for f in t.schema]
    out = G.read_gpu([p], schema, "cpu")
    got = {n: out.columns[n].to_pylist() for n in out.columns}
    for name in t.schema.names:
        exp = _expect_col(t, name)
        g = got[name]
        if name == "d":
            exp = [None if v is None else float(v) for v in exp]
            assert g == pytest.approx(exp)
        elif name == "dt":
            exp = [None if v is None else v for v in exp]
            assert [x if x is None else x.isoformat() for x in g] == \
                [x if x is None else x.isoformat() for x in exp]
        elif name == "f":
            assert g == pytest.approx(exp)
        else:
            assert g == exp, name


def test_sim_decode_column_subset(tmp_path, sim):
    p = str(tmp_path / "sub.parquet")
    t = _write_mixed(p, nrows=500)
    out = G.read_gpu([p], [("s_hi", None), ("k", None)], "cpu")
    assert list(out.columns.keys()) == ["s_hi", "k"]
    assert out.columns["k"].to_pylist() == t.column("k").to_pylist()
    assert out.columns["s_hi"].to_pylist() == t.column("s_hi").to_pylist()


def test_sim_decode_multifile_concat(tmp_path, sim):
    p1 = str(tmp_path / "a.parquet")
    p2 = str(tmp_path / "b.parquet")
    t1 = _write_mixed(p1, nrows=300)
    t2 = _write_mixed(p2, nrows=200)
    out = G.read_gpu([p1, p2], [("k", None), ("s_low", None)], "cpu")
    assert out.columns["k"].to_pylist() == \
        t1.column("k").to_pylist() + t2.column("k").to_pylist()
    assert out.columns["s_low"].to_pylist() == \
        t1.column("s_low").to_pylist() + t2.column("s_low").to_pylist()


def test_sim_dict_string_stays_dict(tmp_path, sim):
    p = str(tmp_path / "d.parquet")
    _write_mixed(p, nrows=400, row_groups=2)
    out = G.read_gpu([p], [("s_low", None)], "cpu")
    col = out.columns["s_low"]
    assert col.is_dict  # RLE_DICTIONARY pages land as dict StringColumn


def test_plain_bytearray_path(tmp_path, sim):
    p = str(tmp_path / "pb.parquet")
    vals = [f"value-{i}-{'y'*(i % 17)}" for i in range(2000)]
    t = pa.table({"s": pa.array(vals)})
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_size=512, data_page_version="1.0")
    out = G.read_gpu([p], [("s", None)], "cpu")
    assert out.columns["s"].to_pylist() == vals


def test_timestamp_units(tmp_path, sim):
    import datetime

    p = str(tmp_path / "ts.parquet")
    base = datetime.datetime(2021, 5, 4, 12, 30, 1, 500000)
    vals = [base + datetime.timedelta(seconds=i) for i in range(100)]
    t = pa.table({"ts": pa.array(vals, pa.timestamp("ms"))})
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_version="1.0")
    out = G.read_gpu([p], [("ts", None)], "cpu")
    got = out.columns["ts"].data.tolist()  # engine timestamps: epoch micros
    import calendar

    exp = [int(calendar.timegm(v.timetuple())) * 1_000_000 + v.microsecond
           for v in vals]
    assert got == exp


def test_tpch_shard_sim_decode(tmp_path, sim):
    """The exact files the scan-inclusive bench writes decode correctly via
    the kernel contracts (simulator) — covers DELTA ints, FLBA decimals,
    dict strings, DELTA_LENGTH comments at datagen encodings."""
    import sail_amd
    from sail_amd.datagen.tpch import TpchGenerator, write_tpch_parquet

    gen = TpchGenerator(sf=0.003, device="cpu", seed=1, rank=0, world=1)
    tables = gen.generate_all()
    paths = write_tpch_parquet(tables, str(tmp_path / "shards"))
    for name in ("lineitem", "orders", "customer"):
        t = pq.read_table(paths[name])
        out = G.read_gpu([paths[name]],
                         [(f.name, None) for f in t.schema], "cpu")
        for cn in t.schema.names:
            exp = t.column(cn).to_pylist()
            got = out.columns[cn].to_pylist()
            if isinstance(exp[0], float) or str(t.schema.field(cn).type).startswith("decimal"):
                exp = [None if v is None else float(v) for v in exp]
                assert got == pytest.approx(exp), (name, cn)
            elif str(t.schema.field(cn).type) == "date32[day]":
                assert [str(x) for x in got] == [str(x) for x in exp], (name, cn)
            else:
                assert got == exp, (name, cn)


def test_tpch_scan_mode_matches_resident(tmp_path):
    """register_tpch_parquet (scan views + pruned DataSourceRead leaves)
    returns the same rows as the resident tables for a query mix."""
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch, register_tpch_parquet
    from sail_amd.datagen.tpch_queries import QUERIES

    s1 = sail_amd.SessionContext(device="cpu")
    register_tpch(s1, sf=0.005, device="cpu")
    s2 = sail_amd.SessionContext(device="cpu")
    register_tpch_parquet(s2, sf=0.005, device="cpu",
                          data_dir=str(tmp_path / "d"))
    for q in (1, 3, 6, 13, 17, 21):
        assert s1.sql(QUERIES[q]).collect() == s2.sql(QUERIES[q]).collect(), q


def test_scanned_dicts_are_lex_sorted(tmp_path, sim):
    """Parquet dictionary pages arrive in writer first-occurrence order;
    the scan must restore the engine invariant code order == byte order
    (StringColumn min/max, comparisons and ORDER BY all compare codes)."""
    p = str(tmp_path / "d1.parquet")
    vals = ["zebra", "apple", "mango", "zebra", "apple", "banana"] * 100
    t = pa.table({"s": pa.array(vals)})
    pq.write_table(t, p, use_dictionary=["s"], compression="none")
    out = G.read_gpu([p], [("s", None)], "cpu")
    c = out.columns["s"]
    assert c.is_dict
    dv = c.dict_values()
    assert dv == sorted(dv)
    assert c.to_pylist() == vals

    # multi-row-group: merged dictionary must be sorted too, and the
    # cached re-read must agree
    p2 = str(tmp_path / "d2.parquet")
    vals2 = ["zz", "mm", "aa"] * 200 + ["qq", "bb", "zz"] * 200
    pq.write_table(pa.table({"s": pa.array(vals2)}), p2,
                   use_dictionary=["s"], compression="none",
                   row_group_size=600)
    for _ in range(2):  # second pass exercises _DICT_CACHE
        out2 = G.read_gpu([p2], [("s", None)], "cpu")
        dv2 = out2.columns["s"].dict_values()
        assert dv2 == sorted(dv2)
        assert out2.columns["s"].to_pylist() == vals2


def test_lex_perm_edge_cases():
    """_lex_perm: prefixes sort first, empties first, bytes >= 0x80 sort
    after ASCII (unsigned byte order), long common prefixes break ties."""
    import torch

    def mk(strs):
        bs = [s.encode() if isinstance(s, str) else s for s in strs]
        offs = torch.zeros(len(bs) + 1, dtype=torch.int64)
        offs[1:] = torch.cumsum(
            torch.tensor([len(b) for b in bs], dtype=torch.int64), 0)
        blob = torch.tensor(list(b"".join(bs)), dtype=torch.uint8)
        return offs, blob, bs

    cases = [
        ["http://a/xyz", "http://a/x", "http://a/xy", "", "http://a"],
        [b"a", b"\xff", b"\x80", b"z", b"\x7f"],
        ["same-long-prefix-0123456789-b", "same-long-prefix-0123456789-a"],
        ["", "", "a"],
    ]
    for strs in cases:
        offs, blob, bs = mk(strs)
        perm = G._lex_perm(offs, blob)
        got = [bs[i] for i in perm.tolist()]
        assert got == sorted(bs), strs


def test_scan_min_max_over_dict_strings(tmp_path, sim):
    """End-to-end: MIN/MAX over a scanned dict column (order-dependent)."""
    import sail_amd

    p = str(tmp_path / "mm.parquet")
    vals = (["walnut", "cherry", "fig"] * 50) + (["apricot", "plum"] * 30)
    keys = [i % 2 for i in range(len(vals))]
    pq.write_table(pa.table({"s": pa.array(vals), "k": pa.array(keys)}), p,
                   use_dictionary=["s"], compression="none")
    exp = {}
    for k, v in zip(keys, vals):
        lo, hi = exp.get(k, (v, v))
        exp[k] = (min(lo, v), max(hi, v))
    s = sail_amd.SessionContext(device="cpu")
    s.sql(f"CREATE TEMP VIEW t AS SELECT * FROM parquet.`{p}`")
    rows = s.sql("SELECT k, MIN(s), MAX(s) FROM t GROUP BY k ORDER BY k"
                 ).collect()
    assert rows == [(k, exp[k][0], exp[k][1]) for k in sorted(exp)]
