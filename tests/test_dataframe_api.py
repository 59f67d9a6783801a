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


class TestPySparkParityBatch:
    """The df method surface added for PySpark parity (round 2):
    set ops, sampling, stat/na helpers, rollup/cube/pivot, misc."""

    @pytest.fixture()
    def df(self, session):
        session.create_dataframe(
            {"k": ["a", "b", "a", "b"], "v": [1, 2, 3, 4],
             "g": ["x", "x", "y", "y"]}, name="pp_t")
        return session.table("pp_t")

    def test_set_ops_and_dedup(self, df):
        assert df.subtract(df).count() == 0
        assert df.exceptAll(df.limit(1)).count() == 3
        assert df.intersectAll(df).count() == 4
        assert df.unionByName(df.select("v", "k", "g").select("k", "v", "g")
                              ).count() == 8
        assert df.dropDuplicates(["k"]).count() == 2
        assert df.unionByName(df.select("k", "v"),
                              allowMissingColumns=True).count() == 8

    def test_sampling_and_splits(self, df):
        assert 0 <= df.sample(0.5, seed=1).count() <= 4
        a, b = df.randomSplit([0.5, 0.5], seed=3)
        assert a.count() + b.count() == 4

    def test_stat_and_na(self, df):
        assert df.stat.corr("v", "v") == pytest.approx(1.0)
        assert df.stat.cov("v", "v") > 0
        assert df.stat.approxQuantile("v", [0.0, 1.0]) == [1.0, 4.0]
        ct = df.stat.crosstab("k", "g").collect()
        assert ct == [("a", 1, 1), ("b", 1, 1)]
        assert df.stat.freqItems(["k"], 0.4) == [["a", "b"]]
        assert df.na.fill(0).count() == 4
        assert df.stat.sampleBy("k", {"a": 1.0}, seed=1).count() == 2

    def test_rollup_cube_pivot_unpivot(self, df):
        assert sorted(df.rollup("k").count().collect(), key=str) == \
            [("a", 2), ("b", 2), (None, 4)]
        assert len(df.cube("k", "g").count().collect()) == 9
        assert df.groupBy("k").pivot("g").agg("sum(v)").collect() == \
            [("a", 1, 3), ("b", 2, 4)]
        assert df.groupBy("k").pivot("g", ["x"]).agg("sum(v)"
                                                     ).collect() == \
            [("a", 1), ("b", 2)]
        assert df.select("k", "v").unpivot(
            ["k"], ["v"], "var", "val").count() == 4

    def test_misc_surface(self, df):
        assert df.dtypes == [("k", "string"), ("v", "int"),
                             ("g", "string")]
        assert df.toDF("a", "b", "c").columns == ["a", "b", "c"]
        assert df.crossJoin(df.alias("d2")).count() == 16
        assert df.orderBy("v").tail(2) == [("a", 3, "y"), ("b", 4, "y")]
        assert df.orderBy("v").offset(3).count() == 1
        assert df.filter("v > 100").isEmpty()
        assert not df.isEmpty()
        cached = df.cache()
        assert cached.count() == 4 and cached.count() == 4
        assert df.colRegex("`[kg]`") == ["k", "g"]
        assert df.withColumns({"w": "v*2"}).columns[-1] == "w"
        assert df.transform(lambda d: d.limit(1)).count() == 1
        assert df.hint("broadcast").count() == 4
        got = []
        df.foreach(got.append)
        assert len(got) == 4
