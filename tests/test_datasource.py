"""Parquet/CSV/JSON read & write round-trips (host decode path)."""
import os

import pytest

import sail_amd
from sail_amd.engine import types as T


@pytest.fixture()
def s(tmp_path):
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe(
        {"a": [1, 2, 3, 4], "b": [1.5, 2.5, None, 4.0], "c": ["x", "yy", "zzz", None],
         "d": ["2024-01-01", "2023-06-30", "2022-03-15", "2021-12-31"]},
        schema={"a": T.I64, "b": T.F64, "c": T.STRING, "d": T.DATE}, name="t")
    return s


def test_parquet_roundtrip(s, tmp_path):
    p = str(tmp_path / "out_pq")
    s.table("t").write.mode("overwrite").parquet(p)
    df = s.read.parquet(p)
    assert sorted(df.collect()) == sorted(s.table("t").collect())
    # SQL direct path read
    rows = s.sql(f"SELECT count(*), sum(a) FROM parquet.`{p}`").collect()
    assert rows == [(4, 10)]


def test_csv_roundtrip(s, tmp_path):
    p = str(tmp_path / "out_csv")
    s.sql("SELECT a, c FROM t WHERE c IS NOT NULL").write.format("csv").save(p)
    df = s.read.csv(p)
    assert sorted(df.collect()) == [(1, "x"), (2, "yy"), (3, "zzz")]


def test_json_read(s, tmp_path):
    p = tmp_path / "data.jsonl"
    p.write_text('{"x": 1, "y": "a"}\n{"x": 2, "y": "b"}\n')
    df = s.read.json(str(p))
    assert sorted(df.collect()) == [(1, "a"), (2, "b")]


def test_parquet_predicate_after_read(s, tmp_path):
    p = str(tmp_path / "pq2")
    s.table("t").write.mode("overwrite").parquet(p)
    rows = s.sql(f"SELECT a FROM parquet.`{p}` WHERE d >= DATE '2023-01-01' ORDER BY a").collect()
    assert rows == [(1,), (2,)]


def test_text_format(session, tmp_path):
    p = tmp_path / "a.txt"
    p.write_text("hello\nworld\n")
    df = session.read.format("text").load(str(p))
    assert df.collect() == [("hello",), ("world",)]
    out = tmp_path / "out"
    df.write.format("text").save(str(out))
    assert session.read.format("text").load(str(out)).collect() == [
        ("hello",), ("world",)]


def test_binary_file_format(session, tmp_path):
    p = tmp_path / "b.bin"
    p.write_bytes(b"\x00\x01binary")
    rows = session.read.format("binaryFile").load(str(p)).collect()
    assert rows[0][1] == 8


def test_arrow_format_roundtrip(session, tmp_path):
    d = str(tmp_path / "ar")
    df = session.create_dataframe({"x": [1, 2], "y": ["a", "b"]})
    df.write.format("arrow").save(d)
    assert session.read.format("arrow").load(d).collect() == [(1, "a"), (2, "b")]


def test_hive_partitioned_parquet(session, tmp_path):
    import os

    import pyarrow as pa
    import pyarrow.parquet as pq

    d = str(tmp_path / "part")
    for y in (2023, 2024):
        for m in ("01", "02"):
            sub = os.path.join(d, f"year={y}", f"month={m}")
            os.makedirs(sub)
            pq.write_table(pa.table({"v": [y * 100 + int(m)]}),
                           os.path.join(sub, "p.parquet"))
    df = session.read.format("parquet").load(d)
    assert [n for n, _ in df.plan.schema] == ["v", "year", "month"]
    assert sorted(df.collect()) == [
        (202301, 2023, 1), (202302, 2023, 2), (202401, 2024, 1), (202402, 2024, 2)]
    assert session.sql(
        f"SELECT v FROM parquet.`{d}` WHERE year = 2024 AND month = 2").collect() == [
        (202402,)]


def test_partition_pruning_skips_files(session, tmp_path, monkeypatch):
    import os

    import pyarrow as pa
    import pyarrow.parquet as pq

    import sail_amd.datasource.parquet_io as pio

    d = str(tmp_path / "pp")
    for y in (2023, 2024):
        sub = os.path.join(d, f"year={y}")
        os.makedirs(sub)
        pq.write_table(pa.table({"v": [y]}), os.path.join(sub, "p.parquet"))
    calls = []
    orig = pio.pq.read_table

    def spy(f, *a, **k):
        calls.append(str(f))
        return orig(f, *a, **k)

    monkeypatch.setattr(pio.pq, "read_table", spy)
    assert session.sql(f"SELECT sum(v) FROM parquet.`{d}` WHERE year = 2023").collect() == [
        (2023,)]
    read_dirs = {os.path.basename(os.path.dirname(c)) for c in calls
                 if c.endswith(".parquet")}
    assert read_dirs == {"year=2023"}


def test_partitioned_parquet_write_roundtrip(session, tmp_path):
    import os

    d = str(tmp_path / "wout")
    df = session.create_dataframe({"year": [2023, 2023, 2024], "v": [1, 2, 3]})
    w = df.write.format("parquet")
    (w.partition_by if hasattr(w, "partition_by") else w.partitionBy)("year").save(d)
    assert sorted(os.listdir(d)) == ["year=2023", "year=2024"]
    assert sorted(session.read.format("parquet").load(d).collect()) == [
        (1, 2023), (2, 2023), (3, 2024)]
    assert session.sql(f"SELECT sum(v) FROM parquet.`{d}` WHERE year = 2023").collect() == [
        (3,)]
