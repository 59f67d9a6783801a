"""PySpark-style DataFrame composition (SQL-backed chaining)."""
import pytest

import sail_amd


@pytest.fixture()
def df():
    s = sail_amd.SessionContext(device="cpu")
    return s.create_dataframe({"a": [1, 2, 3], "g": ["x", "x", "y"]})


def test_select_filter(df):
    assert df.filter("a > 1").select("a").collect() == [(2,), (3,)]
    assert df.where("g = 'y'").collect() == [(3, "y")]


def test_with_column_order_limit(df):
    rows = df.with_column("b", "a * 10").order_by("a DESC").limit(2).collect()
    assert rows == [(3, "y", 30), (2, "x", 20)]
    assert df.with_column_renamed("a", "aa").columns == ["aa", "g"]


def test_group_agg_join_union(df):
    assert df.group_by("g").agg("sum(a) AS s").order_by("g").collect() == [
        ("x", 3), ("y", 3)]
    assert df.group_by("g").count().order_by("g").collect() == [("x", 2), ("y", 1)]
    other = df.session.create_dataframe({"a": [2, 3], "z": ["p", "q"]})
    assert df.join(other, "a").order_by("a").collect() == [
        (2, "x", "p"), (3, "y", "q")]
    assert df.join(other, "a", how="anti").collect() == [(1, "x")]
    assert df.select("a").union(other.select("a")).distinct().order_by("a").collect() == [
        (1,), (2,), (3,)]


def test_misc(df):
    assert df.drop("g").columns == ["a"]
    assert df.first() == (1, "x")
    assert len(df.head(2)) == 2
    assert df.agg("max(a)").collect() == [(3,)]
