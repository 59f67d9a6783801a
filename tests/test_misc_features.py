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
