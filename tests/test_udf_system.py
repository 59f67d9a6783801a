"""Python UDFs, system tables, config (ref: sail-python-udf, sail-catalog-system)."""
import sail_amd
from sail_amd.engine import types as T


def test_python_udf():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1, 2, 3]}, name="t")
    s.udf.register("plus_ten", lambda x: x + 10, "bigint")
    assert s.sql("SELECT plus_ten(a) FROM t ORDER BY a").collect() == [(11,), (12,), (13,)]


def test_python_udf_string():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1, 2]}, name="t")
    s.udf.register("tag", lambda x: f"row-{x}", "string")
    assert s.sql("SELECT tag(a) FROM t ORDER BY a").collect() == [("row-1",), ("row-2",)]


def test_udf_two_args():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1, 2], "b": [10, 20]}, name="t")
    s.udf.register("addxy", lambda x, y: x * y, "bigint")
    assert s.sql("SELECT addxy(a, b) FROM t ORDER BY a").collect() == [(10,), (40,)]


def test_system_queries_table():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1]}, name="t")
    s.sql("SELECT a FROM t").collect()
    rows = s.sql("SELECT query, rows FROM system_queries").collect()
    assert any("SELECT a FROM t" in q for q, _ in rows)


def test_system_tables_table():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1, 2]}, name="mytable")
    rows = dict(s.sql("SELECT tableName, rows FROM system_tables").collect())
    assert rows.get("mytable") == 2


def test_config_defaults_and_env(monkeypatch):
    monkeypatch.setenv("SAIL_OPTIMIZER_ENABLE_JOIN_REORDER", "false")
    s = sail_amd.SessionContext(device="cpu")
    assert s.conf["sail.optimizer.enable.join.reorder"] == "false"
    assert s.conf["spark.sql.session.timeZone"] == "UTC"


def test_udaf_register_and_group(session):
    s = session
    import math

    s.udf.register_aggregate(
        "geo_mean",
        lambda vals: math.exp(sum(math.log(v) for v in vals) / len(vals)) if vals else None,
        "double")
    s.create_dataframe({"g": ["a", "a", "b"], "v": [2.0, 8.0, 5.0]}, name="ua")
    rows = s.sql("SELECT g, geo_mean(v) FROM ua GROUP BY g ORDER BY g").collect()
    assert rows[0][0] == "a" and abs(rows[0][1] - 4.0) < 1e-9
    assert rows[1][0] == "b" and abs(rows[1][1] - 5.0) < 1e-9
    # global aggregate
    assert abs(s.sql("SELECT geo_mean(v) FROM ua").collect()[0][0] - 4.30886938) < 1e-6


def test_python_data_source(session, tmp_path):
    s = session
    """User-defined format (ref: sail-data-source formats/python/)."""
    from sail_amd.datasource.registry import register_format
    from sail_amd.engine.column import Column, Table
    from sail_amd.engine import types as T

    class FibSource:
        def infer_schema(self, paths, options):
            return [("n", T.I64), ("fib", T.I64)]

        def read(self, paths, schema, device, options):
            k = int(options.get("count", "8"))
            fibs = [0, 1]
            while len(fibs) < k:
                fibs.append(fibs[-1] + fibs[-2])
            return Table({"n": Column.from_values(list(range(k)), T.I64, device=device),
                          "fib": Column.from_values(fibs[:k], T.I64, device=device)})

    register_format("fib", FibSource())
    df = s.read.format("fib").option("count", 7).load("ignored")
    assert df.collect() == [(0, 0), (1, 1), (2, 1), (3, 2), (4, 3), (5, 5), (6, 8)]


def test_udtf_table_function(session):
    s = session

    def primes_upto(n):
        out = []
        for x in range(2, n + 1):
            if all(x % d for d in range(2, int(x ** 0.5) + 1)):
                out.append(x)
        return {"p": out}

    s.udf.register_table_function("primes", primes_upto, {"p": "bigint"})
    assert s.sql("SELECT * FROM primes(20)").collect() == [
        (2,), (3,), (5,), (7,), (11,), (13,), (17,), (19,)]
    assert s.sql("SELECT sum(p) FROM primes(10) WHERE p > 2").collect() == [(15,)]
    rows = s.sql("SELECT t.p FROM primes(5) t JOIN primes(7) q ON t.p = q.p "
                 "ORDER BY 1").collect()
    assert rows == [(2,), (3,), (5,)]
