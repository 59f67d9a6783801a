"""Negative-path coverage: structured errors, not silent wrong answers."""
import pytest

import sail_amd
from sail_amd import AnalysisException, ParseException


@pytest.fixture()
def s():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"a": [1, 2], "c": ["x", "y"]}, name="t")
    return s


def test_parse_errors(s):
    for bad in ["SELEC 1", "SELECT * FROM", "SELECT a FROM t WHERE",
                "SELECT a FROM t GROUP", "MERGE INTO t"]:
        with pytest.raises(ParseException):
            s.sql(bad)
    # `SELECT FROM t` surfaces at analysis (FROM parses as a column ref);
    # either structured error class is acceptable
    with pytest.raises(sail_amd.SailError):
        s.sql("SELECT FROM t")


def test_analysis_errors(s):
    with pytest.raises(AnalysisException):
        s.sql("SELECT nope FROM t")
    with pytest.raises(AnalysisException):
        s.sql("SELECT * FROM missing_table")
    with pytest.raises(AnalysisException):
        s.sql("SELECT not_a_function(a) FROM t")
    with pytest.raises(AnalysisException):
        s.sql("SELECT t2.a FROM t")  # unknown qualifier
    with pytest.raises(AnalysisException):
        s.sql("INSERT INTO missing_table VALUES (1)")
    with pytest.raises(AnalysisException):
        s.sql("SELECT a FROM t ORDER BY 5")  # ordinal out of range


def test_ambiguous_column(s):
    s.create_dataframe({"a": [1]}, name="u")
    with pytest.raises(AnalysisException, match="ambiguous|duplicate"):
        s.sql("SELECT a FROM t JOIN u ON t.a = u.a WHERE a > 0 AND t.c = 'x'")


def test_scalar_subquery_multirow(s):
    with pytest.raises(Exception, match="more than one row"):
        s.sql("SELECT (SELECT a FROM t)").collect()


def test_empty_table_paths(s):
    s.sql("CREATE TABLE e (x INT, y STRING)")
    assert s.sql("SELECT count(*), sum(x), max(y) FROM e").collect() == [(0, None, None)]
    assert s.sql("SELECT x FROM e ORDER BY x LIMIT 5").collect() == []
    assert s.sql("SELECT x, count(*) FROM e GROUP BY x").collect() == []
    assert s.sql("SELECT * FROM t JOIN e ON t.a = e.x").collect() == []
    assert s.sql("SELECT * FROM t LEFT JOIN e ON t.a = e.x ORDER BY a").collect() == [
        (1, "x", None, None), (2, "y", None, None)]


def test_division_semantics(s):
    # Spark: x / 0 -> null (non-ANSI)
    assert s.sql("SELECT a / 0 FROM t ORDER BY a").collect() == [(None,), (None,)]
    assert s.sql("SELECT a % 0 FROM t ORDER BY a").collect() == [(None,), (None,)]


def test_outer_joins_empty_sides(s):
    s.sql("CREATE TABLE e2 (x INT, y STRING)")
    assert s.sql("SELECT * FROM e2 RIGHT JOIN t ON e2.x = t.a ORDER BY a").collect() == [
        (None, None, 1, "x"), (None, None, 2, "y")]
    assert s.sql("SELECT * FROM e2 FULL JOIN t ON e2.x = t.a ORDER BY a").collect() == [
        (None, None, 1, "x"), (None, None, 2, "y")]
    assert s.sql("SELECT * FROM t FULL JOIN e2 ON t.a = e2.x ORDER BY a").collect() == [
        (1, "x", None, None), (2, "y", None, None)]
