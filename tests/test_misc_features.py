"""TABLESAMPLE, df.checkpoint(), vectorized UDFs."""
import pytest

import sail_amd
from sail_amd.engine import types as T


@pytest.fixture()
def s():
    return sail_amd.SessionContext(device="cpu")


def test_tablesample_percent_repeatable(s):
    a = s.sql("SELECT count(*) FROM range(10000) TABLESAMPLE (10 PERCENT) REPEATABLE (7) t").collect()
    b = s.sql("SELECT count(*) FROM range(10000) TABLESAMPLE (10 PERCENT) REPEATABLE (7) t").collect()
    assert a == b
    assert 800 < a[0][0] < 1200


def test_tablesample_rows(s):
    assert s.sql("SELECT count(*) FROM range(100) TABLESAMPLE (5 ROWS) t").collect() == [(5,)]
    # fewer rows than requested: all pass through
    assert s.sql("SELECT count(*) FROM range(3) TABLESAMPLE (5 ROWS) t").collect() == [(3,)]


def test_tablesample_on_subquery(s):
    rows = s.sql("SELECT count(*) FROM (SELECT id FROM range(100) WHERE id < 50) "
                 "TABLESAMPLE (10 ROWS) t").collect()
    assert rows == [(10,)]


def test_checkpoint_truncates_lineage(s):
    s.create_dataframe({"k": ["a", "b", "a"], "v": [1, 2, 3]}, name="t")
    df = s.sql("SELECT k, sum(v) AS sv FROM t GROUP BY k")
    cp = df.checkpoint()
    # the checkpointed plan is a plain parquet read
    assert "DataSourceRead" in cp.explain() or "parquet" in cp.explain().lower()
    assert sorted(cp.collect()) == [("a", 4), ("b", 2)]
    # mutating the base table does not affect the checkpoint
    s.create_dataframe({"k": ["z"], "v": [100]}, name="t")
    assert sorted(cp.collect()) == [("a", 4), ("b", 2)]


def test_vectorized_udf(s):
    import numpy as np

    calls = []

    def plus_one(arr):
        calls.append(len(arr))
        return arr + 1

    s.udf.register("vplus", plus_one, "bigint", vectorized=True)
    s.create_dataframe({"v": [1, 2, 3, 4]}, schema={"v": T.I64}, name="u")
    rows = s.sql("SELECT vplus(v) AS r FROM u ORDER BY r").collect()
    assert rows == [(2,), (3,), (4,), (5,)]
    assert calls == [4]  # exactly one batched call, not 4 per-row calls


def test_vectorized_udf_string(s):
    def upper_all(arr):
        return [x.upper() for x in arr]

    s.udf.register("vupper", upper_all, "string", vectorized=True)
    s.create_dataframe({"c": ["ab", "cd"]}, name="u2")
    assert s.sql("SELECT vupper(c) FROM u2 ORDER BY 1").collect() == [("AB",), ("CD",)]


def test_scalar_udf_still_works(s):
    s.udf.register("sq", lambda x: x * x, "bigint")
    s.create_dataframe({"v": [2, 3]}, schema={"v": T.I64}, name="u3")
    assert s.sql("SELECT sq(v) FROM u3 ORDER BY 1").collect() == [(4,), (9,)]


def test_structured_errors(s):
    import sail_amd as sa

    with pytest.raises(sa.ParseException) as ei:
        s.sql("SELEC 1")
    assert ei.value.sql_state == "42601"
    with pytest.raises(sa.AnalysisException):
        s.sql("SELECT * FROM does_not_exist")
    with pytest.raises(sa.AnalysisException):
        s.sql("SELECT no_such_fn_xyz(1)")
    # all are SailError
    with pytest.raises(sa.SailError):
        s.sql("SELEC 1")


def test_explain_analyze(s):
    s.create_dataframe({"a": [1, 2, 3]}, name="ea")
    out = s.sql("EXPLAIN ANALYZE SELECT a, count(*) FROM ea GROUP BY a").collect()[0][0]
    assert "Analyzed (wall times)" in out
    assert "Aggregate" in out and "ms self" in out


def test_system_operators_table(s):
    s.conf["sail.trace"] = "true"
    s.create_dataframe({"a": [1, 2, 3]}, name="sot")
    s.sql("SELECT a, count(*) FROM sot GROUP BY a").collect()
    rows = s.sql("SELECT operator FROM system_operators").collect()
    ops = {r[0] for r in rows}
    assert "Aggregate" in ops and "Read" in ops


def test_show_cache_analyze_commands(s):
    s.create_dataframe({"a": [1, 2, 2]}, name="cat_t")
    fns = s.sql("SHOW FUNCTIONS").collect()
    assert len(fns) > 250
    assert s.sql("SHOW FUNCTIONS LIKE 'array_c%'").collect()[0][0].startswith("array_c")
    assert s.sql("SHOW DATABASES").collect() == [("default",)]
    s.sql("CREATE OR REPLACE TEMP VIEW cv AS SELECT a * 10 AS x FROM cat_t")
    s.sql("CACHE TABLE cv")
    s.create_dataframe({"a": [100]}, name="cat_t")
    assert s.sql("SELECT sum(x) FROM cv").collect() == [(50,)]  # frozen
    s.sql("UNCACHE TABLE cv")
    assert s.sql("SELECT sum(x) FROM cv").collect() == [(1000,)]  # live view again
    rows = s.sql("ANALYZE TABLE cat_t COMPUTE STATISTICS FOR ALL COLUMNS").collect()
    assert rows[0][0] == "a" and rows[0][1] == 1


def test_dataframe_conveniences(s):
    df = s.create_dataframe({"a": [1, 2, None, 4], "b": ["x", None, "y", "x"]})
    desc = dict((r[0], r[1]) for r in df.describe().collect())
    assert desc["count"] == "3" and desc["min"] == "1"
    assert df.fillna(0).collect()[2][0] == 0
    assert df.fillna("zz").collect()[1][1] == "zz"
    assert len(df.dropna().collect()) == 2
    assert df.replace("x", "XX", subset=["b"]).collect()[0][1] == "XX"


def test_json_functions(s):
    s.create_dataframe({"j": ['{"a": 1, "b": {"c": [10, 20]}}', '{"a": "x"}', None]},
                       name="jt")
    rows = s.sql("SELECT get_json_object(j, '$.a'), get_json_object(j, '$.b.c[1]') "
                 "FROM jt").collect()
    assert rows == [("1", "20"), ("x", None), (None, None)]
    assert s.sql("SELECT to_json(named_struct('p', 1, 's', 'q'))").collect() == [
        ('{"p":1,"s":"q"}',)]
    assert s.sql("SELECT to_json(map('k', 5)), to_json(array(1,2))").collect() == [
        ('{"k":5}', "[1,2]")]
    assert "STRUCT<" in s.sql("SELECT schema_of_json('{\"a\": 1}')").collect()[0][0]


def test_string_coalesce(s):
    s.create_dataframe({"b": ["x", None], "c": [None, "q"]}, name="sc")
    assert s.sql("SELECT coalesce(b, c, 'd') FROM sc").collect() == [("x",), ("q",)]


def test_from_json(s):
    s.create_dataframe({"j": ['{"a": 5, "b": "hi"}', "bad", None]}, name="fj")
    rows = s.sql("SELECT from_json(j, 'a INT, b STRING') FROM fj").collect()
    assert rows == [({"a": 5, "b": "hi"},), (None,), (None,)]
    rows = s.sql("SELECT from_json(j, 'a INT, b STRING').a + 1 FROM fj").collect()
    assert rows == [(6,), (None,), (None,)]


def test_alter_table(s):
    s.create_dataframe({"x": [1, 2]}, name="at")
    s.sql("ALTER TABLE at ADD COLUMNS (y INT, z STRING)")
    assert s.sql("SELECT * FROM at ORDER BY x").collect() == [
        (1, None, None), (2, None, None)]
    s.sql("ALTER TABLE at RENAME COLUMN y TO yy")
    s.sql("UPDATE at SET yy = 5 WHERE x = 1")
    assert s.sql("SELECT x, yy FROM at ORDER BY x").collect() == [(1, 5), (2, None)]
    s.sql("ALTER TABLE at DROP COLUMN z")
    assert [r[0] for r in s.sql("DESCRIBE at").collect()] == ["x", "yy"]
    s.sql("ALTER TABLE at RENAME TO at2")
    assert s.sql("SELECT count(*) FROM at2").collect() == [(2,)]


def test_window_tumbling_group_by(s):
    s.create_dataframe(
        {"ts": ["2024-01-01 00:05:00", "2024-01-01 00:55:00", "2024-01-01 01:10:00"],
         "v": [1, 2, 30]}, name="win_ev")
    rows = s.sql(
        "SELECT window(to_timestamp(ts), '1 hour').start AS ws, sum(v) "
        "FROM win_ev GROUP BY window(to_timestamp(ts), '1 hour') ORDER BY 1").collect()
    assert rows == [(1704067200000000, 3), (1704070800000000, 30)]


def test_window_by_ordinal_and_end(s):
    s.create_dataframe({"ts": ["2024-01-01 00:05:00", "2024-01-01 00:55:00"]},
                         name="win_ev2")
    rows = s.sql(
        "SELECT window(to_timestamp(ts), '30 minutes').end AS we, count(*) "
        "FROM win_ev2 GROUP BY 1 ORDER BY 1").collect()
    assert rows == [(1704069000000000, 1), (1704070800000000, 1)]


def test_window_time(s):
    s.create_dataframe({"ts": ["2024-01-01 00:05:00"]}, name="win_ev3")
    rows = s.sql(
        "SELECT window_time(window(to_timestamp(ts), '1 hour')) FROM win_ev3").collect()
    assert rows == [(1704070799999999,)]


def test_otlp_file_export(tmp_path, monkeypatch):
    """Telemetry export (ref: sail-telemetry OTLP): traced queries append
    OTLP/JSON ResourceSpans + ResourceMetrics lines."""
    import json

    import sail_amd

    out = str(tmp_path / "otel" / "trace.jsonl")
    monkeypatch.setenv("SAIL_TRACE", "1")
    monkeypatch.setenv("SAIL_OTEL_FILE", out)
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1, 2, 3]}, name="otel_t")
    assert s.sql("SELECT sum(a) FROM otel_t WHERE a > 1").collect() == [(5,)]
    lines = [json.loads(l) for l in open(out)]
    spans_line = next(l for l in lines if "resourceSpans" in l)
    metrics_line = next(l for l in lines if "resourceMetrics" in l)
    spans = spans_line["resourceSpans"][0]["scopeSpans"][0]["spans"]
    names = {sp["name"] for sp in spans}
    assert "ExecutePlan" in names and "Aggregate" in names
    root = next(sp for sp in spans if sp["name"] == "ExecutePlan")
    children = [sp for sp in spans if sp.get("parentSpanId") == root["spanId"]]
    assert children  # operator spans nest under the query span
    ms = metrics_line["resourceMetrics"][0]["scopeMetrics"][0]["metrics"]
    mnames = {m["name"] for m in ms}
    assert {"execution.output_row_count", "execution.elapsed_compute_time",
            "session.query_count"} <= mnames


def test_variant_get_typed(session):
    """variant_get with a literal type arg returns that type (Spark
    semantics), not the JSON text."""
    q = session.sql
    assert q("SELECT variant_get(parse_json('{\"a\":1}'), '$.a', 'int')"
             ).collect() == [(1,)]
    assert q("SELECT variant_get(parse_json('{\"a\":1.5}'), '$.a', "
             "'double')").collect() == [(1.5,)]
    assert q("SELECT variant_get(parse_json('{\"a\":2}'), '$.a', 'bigint')"
             " + 1").collect() == [(3,)]
    # no type arg: JSON text form
    assert q("SELECT variant_get(parse_json('{\"a\":1}'), '$.a')"
             ).collect() == [("1",)]


def test_insert_column_list(session, tmp_path):
    """INSERT INTO t (cols) VALUES: named columns by position, rest NULL."""
    base = str(tmp_path / "ins")
    session.create_dataframe({"a": [1], "b": ["x"], "c": [1.5]},
                             name="ins_src")
    session.sql(f"CREATE TABLE delta.`{base}` AS SELECT * FROM ins_src")
    session.sql(f"INSERT INTO delta.`{base}` (c, a) VALUES (9.5, 7)")
    assert session.sql(f"SELECT * FROM delta.`{base}` ORDER BY a"
                       ).collect() == [(1, "x", 1.5), (7, None, 9.5)]
    # parenthesized-query INSERT is still a query, not a column list
    session.sql(f"INSERT INTO delta.`{base}` (SELECT 2, 'y', 0.5)")
    assert session.sql(f"SELECT count(*) FROM delta.`{base}`"
                       ).collect() == [(3,)]


def test_quantified_subquery_comparisons(session):
    session.sql("CREATE TEMP VIEW qt AS SELECT * FROM VALUES "
                "('a', 1), ('b', 2), ('a', 3) AS t(k, v)")
    q = session.sql
    assert q("SELECT v FROM qt WHERE v > ALL (SELECT v FROM qt "
             "WHERE k = 'b') ORDER BY v").collect() == [(3,)]
    assert q("SELECT v FROM qt WHERE v >= ANY (SELECT v FROM qt "
             "WHERE k = 'b') ORDER BY v").collect() == [(2,), (3,)]
    # empty subquery: ALL is vacuously true, ANY is false
    assert q("SELECT count(*) FROM qt WHERE v > ALL (SELECT v FROM qt "
             "WHERE k = 'z')").collect() == [(3,)]
    assert q("SELECT count(*) FROM qt WHERE v > ANY (SELECT v FROM qt "
             "WHERE k = 'z')").collect() == [(0,)]
    assert q("SELECT v FROM qt WHERE v < SOME (SELECT v FROM qt) "
             "ORDER BY v").collect() == [(1,), (2,)]


def test_lateral_projection(session):
    """LATERAL (SELECT exprs) — the DecorrelateLateralProjection case
    (ref: sail-logical-optimizer)."""
    session.sql("CREATE TEMP VIEW lt AS SELECT * FROM VALUES "
                "('a', 1), ('b', 2) AS t(k, v)")
    q = session.sql
    assert q("SELECT x.w FROM lt, LATERAL (SELECT lt.v + 1 AS w) x "
             "ORDER BY x.w").collect() == [(2,), (3,)]
    assert q("SELECT lt.v, x.w, x.z FROM lt, "
             "LATERAL (SELECT v * 10 AS w, k || '!' AS z) x "
             "ORDER BY v").collect() == [(1, 10, "a!"), (2, 20, "b!")]
    assert q("SELECT v, doubled FROM lt, LATERAL (SELECT v * 2) "
             "AS x(doubled) ORDER BY v").collect() == [(1, 2), (2, 4)]


def test_utility_ddl_surface(session, tmp_path):
    """SHOW COLUMNS/CREATE TABLE/VIEWS/PARTITIONS/TBLPROPERTIES/CATALOGS,
    USE + CREATE/DROP DATABASE, DESCRIBE QUERY, COMMENT ON, REFRESH,
    TRUNCATE (ref: sail-catalog command surface)."""
    s = session
    s.create_dataframe({"k": ["a"], "v": [1]}, name="ddl_t")
    q = s.sql
    assert q("SHOW COLUMNS IN ddl_t").collect() == [("k",), ("v",)]
    assert "CREATE TABLE ddl_t" in q("SHOW CREATE TABLE ddl_t"
                                     ).collect()[0][0]
    assert q("SHOW CATALOGS").collect() == [("spark_catalog",)]
    q("CREATE DATABASE mydb")
    assert ("mydb",) in q("SHOW DATABASES").collect()
    q("USE mydb")
    q("USE default")
    q("DROP DATABASE mydb")
    assert ("mydb",) not in q("SHOW DATABASES").collect()
    q("CREATE DATABASE IF NOT EXISTS default")  # no error
    assert q("DESCRIBE QUERY SELECT 1 AS a").collect() == \
        [("a", "int", "")]
    q("COMMENT ON TABLE ddl_t IS 'demo'")
    assert ("comment", "demo") in q("SHOW TBLPROPERTIES ddl_t").collect()
    q("COMMENT ON TABLE ddl_t IS NULL")
    assert q("SHOW TBLPROPERTIES ddl_t").collect() == []
    q("CREATE TEMP VIEW ddl_v AS SELECT 1")
    assert any(r[1] == "ddl_v" for r in q("SHOW VIEWS").collect())
    assert q("SHOW PARTITIONS ddl_t").collect() == []
    q("REFRESH TABLE ddl_t")
    q("TRUNCATE TABLE ddl_t")
    assert q("SELECT count(*) FROM ddl_t").collect() == [(0,)]


def test_pyspark_catalog_api(session):
    cat = session.catalog
    session.create_dataframe({"a": [1]}, name="cap_t")
    session.sql("CREATE TEMP VIEW cap_v AS SELECT 1")
    assert "cap_t" in cat.listTables() and "cap_v" in cat.listTables()
    assert cat.tableExists("cap_t") and not cat.tableExists("nope")
    assert cat.listColumns("cap_t") in ([("a", "bigint")], [("a", "int")])
    assert "default" in cat.listDatabases()
    assert cat.databaseExists("default")
    assert cat.currentDatabase() == "default"
    assert "sum" in cat.listFunctions("su*")
    assert cat.functionExists("st_srid")
    assert cat.dropTempView("cap_v") and not cat.dropTempView("cap_v")
