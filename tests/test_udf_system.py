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
