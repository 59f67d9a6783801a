"""End-to-end SQL tests on the CPU path (parser -> resolver -> executor)."""
import math

import pytest

import sail_amd
from sail_amd.engine import types as T


@pytest.fixture()
def s():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe(
        {"a": [1, 2, 3, 4, 5], "b": [10.0, 20.0, 30.0, 40.0, 50.0],
         "c": ["x", "y", "x", "z", None], "d": [None, 2, None, 4, 5]},
        schema={"a": T.I32, "b": T.F64, "c": T.STRING, "d": T.I32},
        name="t")
    return s


def test_select_where(s):
    assert s.sql("SELECT a FROM t WHERE a > 2").collect() == [(3,), (4,), (5,)]


def test_arith_and_alias(s):
    rows = s.sql("SELECT a * 2 + 1 AS x FROM t WHERE a <= 2").collect()
    assert rows == [(3,), (5,)]


def test_division_is_double(s):
    rows = s.sql("SELECT a / 2 FROM t WHERE a = 3").collect()
    assert rows == [(1.5,)]


def test_group_by(s):
    rows = s.sql("SELECT c, sum(a) AS sa, count(*) AS n FROM t GROUP BY c ORDER BY c NULLS LAST").collect()
    assert rows == [("x", 4, 2), ("y", 2, 1), ("z", 4, 1), (None, 5, 1)]


def test_global_agg(s):
    rows = s.sql("SELECT sum(a), min(b), max(b), avg(a), count(d) FROM t").collect()
    assert rows == [(15, 10.0, 50.0, 3.0, 3)]


def test_count_distinct(s):
    rows = s.sql("SELECT count(DISTINCT c) FROM t").collect()
    assert rows == [(3,)]


def test_order_limit_offset(s):
    rows = s.sql("SELECT a FROM t ORDER BY a DESC LIMIT 2").collect()
    assert rows == [(5,), (4,)]
    rows = s.sql("SELECT a FROM t ORDER BY a LIMIT 2 OFFSET 1").collect()
    assert rows == [(2,), (3,)]


def test_order_by_hidden_column(s):
    rows = s.sql("SELECT a FROM t ORDER BY b DESC LIMIT 1").collect()
    assert rows == [(5,)]


def test_nulls_in_filter(s):
    # NULL comparisons are not true
    rows = s.sql("SELECT a FROM t WHERE d > 0").collect()
    assert rows == [(2,), (4,), (5,)]


def test_is_null(s):
    assert s.sql("SELECT a FROM t WHERE d IS NULL").collect() == [(1,), (3,)]
    assert s.sql("SELECT count(*) FROM t WHERE c IS NOT NULL").collect() == [(4,)]


def test_case_when(s):
    rows = s.sql("SELECT CASE WHEN a < 3 THEN 'lo' ELSE 'hi' END AS k FROM t ORDER BY a").collect()
    assert rows == [("lo",), ("lo",), ("hi",), ("hi",), ("hi",)]


def test_in_list(s):
    assert s.sql("SELECT a FROM t WHERE a IN (1, 3, 9)").collect() == [(1,), (3,)]
    assert s.sql("SELECT a FROM t WHERE c IN ('x')").collect() == [(1,), (3,)]


def test_between(s):
    assert s.sql("SELECT a FROM t WHERE a BETWEEN 2 AND 4").collect() == [(2,), (3,), (4,)]
    assert s.sql("SELECT a FROM t WHERE a NOT BETWEEN 2 AND 4").collect() == [(1,), (5,)]


def test_like(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"v": ["apple", "banana", "grape", "pineapple"]}, name="f")
    assert s2.sql("SELECT v FROM f WHERE v LIKE '%apple%'").collect() == [("apple",), ("pineapple",)]
    assert s2.sql("SELECT v FROM f WHERE v LIKE 'gra_e'").collect() == [("grape",)]
    assert s2.sql("SELECT v FROM f WHERE v NOT LIKE '%a%'").collect() == []


def test_joins_inner_left(s):
    s.create_dataframe({"a": [1, 2, 6], "z": ["p", "q", "r"]}, name="u")
    rows = s.sql("SELECT t.a, u.z FROM t JOIN u ON t.a = u.a ORDER BY t.a").collect()
    assert rows == [(1, "p"), (2, "q")]
    rows = s.sql("SELECT t.a, u.z FROM t LEFT JOIN u ON t.a = u.a ORDER BY t.a").collect()
    assert rows == [(1, "p"), (2, "q"), (3, None), (4, None), (5, None)]


def test_join_semi_anti(s):
    s.create_dataframe({"a": [1, 2, 6]}, name="u")
    assert s.sql("SELECT a FROM t LEFT SEMI JOIN u USING (a) ORDER BY a").collect() == [(1,), (2,)]
    assert s.sql("SELECT a FROM t LEFT ANTI JOIN u USING (a) ORDER BY a").collect() == [(3,), (4,), (5,)]


def test_cross_join_count(s):
    assert s.sql("SELECT count(*) FROM t, t t2").collect() == [(25,)]


def test_union(s):
    rows = s.sql("SELECT a FROM t WHERE a <= 2 UNION ALL SELECT a FROM t WHERE a >= 4 ORDER BY a").collect()
    assert rows == [(1,), (2,), (4,), (5,)]
    rows = s.sql("SELECT 1 AS x UNION SELECT 1 UNION SELECT 2 ORDER BY x").collect()
    assert rows == [(1,), (2,)]


def test_distinct(s):
    assert s.sql("SELECT DISTINCT c FROM t WHERE c IS NOT NULL ORDER BY c").collect() == [
        ("x",), ("y",), ("z",)]


def test_cte(s):
    rows = s.sql("WITH big AS (SELECT a FROM t WHERE a > 3) SELECT count(*) FROM big").collect()
    assert rows == [(2,)]


def test_subquery_in_from(s):
    rows = s.sql("SELECT x.a2 FROM (SELECT a * 2 AS a2 FROM t) x WHERE x.a2 > 6 ORDER BY 1").collect()
    assert rows == [(8,), (10,)]


def test_scalar_subquery_uncorrelated(s):
    rows = s.sql("SELECT a FROM t WHERE a > (SELECT avg(a) FROM t) ORDER BY a").collect()
    assert rows == [(4,), (5,)]


def test_having(s):
    rows = s.sql("SELECT c, count(*) AS n FROM t GROUP BY c HAVING count(*) > 1").collect()
    assert rows == [("x", 2)]


def test_group_by_ordinal_and_alias(s):
    rows = s.sql("SELECT c AS k, sum(a) FROM t WHERE c IS NOT NULL GROUP BY 1 ORDER BY 1").collect()
    assert rows == [("x", 4), ("y", 2), ("z", 4)]
    rows = s.sql("SELECT c AS k, sum(a) FROM t WHERE c IS NOT NULL GROUP BY k ORDER BY k").collect()
    assert rows == [("x", 4), ("y", 2), ("z", 4)]


def test_agg_expression_over_groups(s):
    rows = s.sql("SELECT c, sum(a) / count(*) AS r FROM t WHERE c = 'x' GROUP BY c").collect()
    assert rows == [("x", 2.0)]


def test_dates():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"d": ["2024-01-15", "2023-06-30", "2024-12-31"]},
                       schema={"d": T.DATE}, name="dt")
    assert s.sql("SELECT year(d) FROM dt ORDER BY d").collect() == [(2023,), (2024,), (2024,)]
    assert s.sql("SELECT month(d), day(d) FROM dt WHERE year(d) = 2023").collect() == [(6, 30)]
    rows = s.sql("SELECT count(*) FROM dt WHERE d >= DATE '2024-01-01'").collect()
    assert rows == [(2,)]
    rows = s.sql("SELECT count(*) FROM dt WHERE d < DATE '1994-01-01' + INTERVAL '1' YEAR").collect()
    assert rows == [(0,)]


def test_decimal_literals_and_agg():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"p": [10.25, 3.75, 1.00]},
                       schema={"p": T.DecimalType(12, 2)}, name="d")
    assert s.sql("SELECT sum(p) FROM d").collect() == [(15.0,)]
    assert s.sql("SELECT count(*) FROM d WHERE p > 3.50").collect() == [(2,)]
    # decimal * decimal keeps exactness
    rows = s.sql("SELECT sum(p * 2.00) FROM d").collect()
    assert rows == [(30.0,)]


def test_values_clause(s):
    rows = s.sql("SELECT col1, col2 FROM (VALUES (1, 'a'), (2, 'b')) v ORDER BY col1").collect()
    assert rows == [(1, "a"), (2, "b")]


def test_select_without_from(s):
    assert s.sql("SELECT 1 + 1").collect() == [(2,)]


def test_functions(s):
    assert s.sql("SELECT abs(-5), round(2.567, 2), floor(2.9), ceil(2.1)").collect() == [
        (5, 2.57, 2, 3)]
    assert s.sql("SELECT upper('ab'), length('abc'), substring('hello', 2, 3)").collect() == [
        ("AB", 3, "ell")]
    assert s.sql("SELECT coalesce(NULL, 5)").collect() == [(5,)]


def test_range_table(s):
    assert s.sql("SELECT count(*), sum(id) FROM range(10)").collect() == [(10, 45)]


def test_create_view(s):
    s.sql("CREATE OR REPLACE TEMP VIEW v AS SELECT a FROM t WHERE a > 3")
    # view resolution happens at query time
    assert s.sql("SELECT count(*) FROM v").collect() == [(2,)]


def test_explain(s):
    out = s.sql("EXPLAIN SELECT a FROM t WHERE a > 1").collect()
    assert "Filter" in out[0][0]


def test_window_row_number(s):
    rows = s.sql(
        "SELECT a, row_number() OVER (PARTITION BY c ORDER BY a) AS rn FROM t WHERE c = 'x' ORDER BY a"
    ).collect()
    assert rows == [(1, 1), (3, 2)]


def test_rollup_cube_grouping_sets():
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"a": ["x", "x", "y"], "b": ["p", "q", "p"], "v": [1, 2, 3]}, name="g")
    rows = s2.sql("SELECT a, b, sum(v) FROM g GROUP BY ROLLUP(a, b) "
                  "ORDER BY a NULLS LAST, b NULLS LAST").collect()
    assert rows == [("x", "p", 1), ("x", "q", 2), ("x", None, 3),
                    ("y", "p", 3), ("y", None, 3), (None, None, 6)]
    rows = s2.sql("SELECT a, grouping(a), count(*) FROM g GROUP BY CUBE(a) "
                  "ORDER BY a NULLS LAST").collect()
    assert rows == [("x", 0, 2), ("y", 0, 1), (None, 1, 3)]
    rows = s2.sql("SELECT a, b, sum(v) FROM g GROUP BY GROUPING SETS ((a), (b)) "
                  "ORDER BY a NULLS LAST, b NULLS LAST").collect()
    assert rows == [("x", None, 3), ("y", None, 3), (None, "p", 4), (None, "q", 2)]


def test_using_join_dedup_and_key_side(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"a": [1, 2, 3], "x": ["p", "q", "r"]}, name="t")
    s2.create_dataframe({"a": [2, 3, 4], "y": [20, 30, 40]}, name="u")
    assert s2.sql("SELECT * FROM t JOIN u USING (a) ORDER BY a").collect() == [
        (2, "q", 20), (3, "r", 30)]
    # RIGHT: key column comes from the right side (4 present, not null)
    assert s2.sql("SELECT * FROM t RIGHT JOIN u USING (a) ORDER BY a").collect() == [
        (2, "q", 20), (3, "r", 30), (4, None, 40)]
    # FULL: key column coalesced across both sides
    assert s2.sql("SELECT * FROM t FULL JOIN u USING (a) ORDER BY a").collect() == [
        (1, "p", None), (2, "q", 20), (3, "r", 30), (4, None, 40)]
    # qualified refs still resolve through the dedup projection
    assert s2.sql("SELECT u.y, t.x FROM t JOIN u USING (a) ORDER BY a").collect() == [
        (20, "q"), (30, "r")]


def test_pivot(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"g": ["x", "x", "y", "y"], "k": ["a", "b", "a", "b"],
                         "v": [1, 2, 3, 4]}, name="p")
    rows = s2.sql("SELECT * FROM p PIVOT (sum(v) FOR k IN ('a', 'b')) ORDER BY g").collect()
    assert rows == [("x", 1, 2), ("y", 3, 4)]
    df = s2.sql("SELECT * FROM p PIVOT (sum(v) FOR k IN ('a' AS col_a, 'b')) ORDER BY g")
    assert [n for n, _ in df.schema] == ["g", "col_a", "b"]
    # missing combination -> null
    s2.create_dataframe({"g": ["x", "y"], "k": ["a", "b"], "v": [1, 4]}, name="p2")
    rows = s2.sql("SELECT * FROM p2 PIVOT (sum(v) FOR k IN ('a', 'b')) ORDER BY g").collect()
    assert rows == [("x", 1, None), ("y", None, 4)]


def test_unpivot(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"id": [1, 2], "q1": [10, 30], "q2": [20, None]}, name="u")
    rows = s2.sql("SELECT * FROM u UNPIVOT (sales FOR quarter IN (q1, q2)) "
                  "ORDER BY id, quarter").collect()
    # Spark default excludeNulls: (2, q2, None) is dropped
    assert rows == [(1, "q1", 10), (1, "q2", 20), (2, "q1", 30)]


def test_direct_address_aggregate_matches_generic(monkeypatch):
    """High-cardinality dense-key fast path (q18 shape) vs the generic
    group_ids path, bit-for-bit."""
    import sail_amd.engine.aggregates as agg_mod

    s2 = sail_amd.SessionContext(device="cpu")
    import torch

    n = 200_000
    g = torch.Generator().manual_seed(5)
    keys = torch.randint(0, 50_000, (n,), generator=g).tolist()
    vals = torch.randint(-100, 100, (n,), generator=g).tolist()
    flt = [v / 7.0 for v in vals]
    s2.create_dataframe({"k": keys, "v": vals, "f": flt}, name="big")
    q = ("SELECT k, sum(v), count(*), avg(f), count(v) FILTER (WHERE v > 0) "
         "FROM big GROUP BY k ORDER BY k")
    monkeypatch.setattr(agg_mod, "DIRECT_MIN_ROWS", 1 << 60)
    want = s2.sql(q).collect()
    monkeypatch.setattr(agg_mod, "DIRECT_MIN_ROWS", 1)
    got = s2.sql(q).collect()
    assert len(got) == len(want)
    for gr, wr in zip(got, want):
        assert gr[0] == wr[0] and gr[1] == wr[1] and gr[2] == wr[2] and gr[4] == wr[4]
        assert abs(gr[3] - wr[3]) < 1e-9


def test_direct_aggregate_multikey_strings(monkeypatch):
    import sail_amd.engine.aggregates as agg_mod

    s2 = sail_amd.SessionContext(device="cpu")
    import torch

    n = 50_000
    g = torch.Generator().manual_seed(6)
    k1 = torch.randint(0, 300, (n,), generator=g).tolist()
    k2 = [["aa", "bb", "cc"][i] for i in torch.randint(0, 3, (n,), generator=g).tolist()]
    v = torch.randint(0, 10, (n,), generator=g).tolist()
    s2.create_dataframe({"k1": k1, "k2": k2, "v": v}, name="mk")
    q = "SELECT k1, k2, sum(v) FROM mk GROUP BY k1, k2 ORDER BY k1, k2"
    monkeypatch.setattr(agg_mod, "DIRECT_MIN_ROWS", 1 << 60)
    want = s2.sql(q).collect()
    monkeypatch.setattr(agg_mod, "DIRECT_MIN_ROWS", 1)
    got = s2.sql(q).collect()
    assert got == want


def test_extended_aggregates(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"g": ["a", "a", "a", "b"], "v": [1.0, 2.0, 10.0, 5.0],
                         "w": [2.0, 4.0, 20.0, 1.0], "c": ["x", "y", "z", "q"],
                         "i": [12, 10, 6, 7]}, name="ea2")
    assert s2.sql("SELECT approx_count_distinct(v) FROM ea2").collect() == [(4,)]
    r = s2.sql("SELECT corr(v, w), covar_samp(v, w), covar_pop(v, w) FROM ea2").collect()[0]
    assert abs(r[0] - 0.86445398) < 1e-6 and abs(r[1] - 31.1666667) < 1e-5
    assert s2.sql("SELECT g, min_by(c, v), max_by(c, v) FROM ea2 GROUP BY g "
                  "ORDER BY g").collect() == [("a", "x", "z"), ("b", "q", "q")]
    assert s2.sql("SELECT bit_and(i), bit_or(i), bit_xor(i) FROM ea2").collect() == [(0, 15, 7)]
    assert s2.sql("SELECT string_agg(c, '-') FROM ea2").collect() == [("x-y-z-q",)]
    sk, ku = s2.sql("SELECT skewness(v), kurtosis(v) FROM ea2").collect()[0]
    assert abs(sk - 0.62973761) < 1e-6
    assert s2.sql("SELECT percentile(v, 0.25) FROM ea2").collect() == [(1.75,)]
    s2.create_dataframe({"m": [1, 1, 2, 2, 2, 3]}, name="mm2")
    assert s2.sql("SELECT mode(m) FROM mm2").collect() == [(2,)]


def test_regr_aggregates(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"x": [1.0, 2.0, 3.0, 4.0], "y": [2.1, 3.9, 6.1, 8.0]}, name="rg")
    slope, icept, r2, n = s2.sql(
        "SELECT regr_slope(y, x), regr_intercept(y, x), regr_r2(y, x), "
        "regr_count(y, x) FROM rg").collect()[0]
    assert abs(slope - 1.99) < 0.02 and n == 4 and r2 > 0.99


def test_topk_matches_full_sort(monkeypatch):
    import sail_amd.engine.executor as ex
    import torch

    s2 = sail_amd.SessionContext(device="cpu")
    g = torch.Generator().manual_seed(13)
    n = 300_000
    c = torch.randint(0, 500, (n,), generator=g).tolist()   # heavy ties
    k = torch.randint(0, 10**9, (n,), generator=g).tolist()
    s2.create_dataframe({"c": c, "k": k}, name="tk")
    q = "SELECT c, k FROM tk ORDER BY c DESC, k LIMIT 25"
    monkeypatch.setattr(ex, "TOPK_MIN_ROWS", 1 << 60)
    want = s2.sql(q).collect()
    monkeypatch.setattr(ex, "TOPK_MIN_ROWS", 1)
    got = s2.sql(q).collect()
    assert got == want
    # with OFFSET and ascending + nulls
    q2 = "SELECT c FROM tk ORDER BY c LIMIT 10 OFFSET 5"
    monkeypatch.setattr(ex, "TOPK_MIN_ROWS", 1 << 60)
    want2 = s2.sql(q2).collect()
    monkeypatch.setattr(ex, "TOPK_MIN_ROWS", 1)
    assert s2.sql(q2).collect() == want2


def test_named_windows_and_by_all(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"g": ["a", "a", "b"], "v": [1, 2, 3]}, name="nb")
    rows = s2.sql("SELECT g, v, sum(v) OVER w FROM nb WINDOW w AS (PARTITION BY g) "
                  "ORDER BY g, v").collect()
    assert rows == [("a", 1, 3), ("a", 2, 3), ("b", 3, 3)]
    assert s2.sql("SELECT g, sum(v) FROM nb GROUP BY ALL ORDER BY g").collect() == [
        ("a", 3), ("b", 3)]
    assert s2.sql("SELECT v, g FROM nb ORDER BY ALL DESC").collect() == [
        (3, "b"), (2, "a"), (1, "a")]
    assert s2.sql("SELECT * FROM nb SORT BY v DESC").collect()[0] == ("b", 3)
    assert len(s2.sql("SELECT * FROM nb CLUSTER BY g").collect()) == 3
    assert len(s2.sql("SELECT * FROM nb DISTRIBUTE BY g").collect()) == 3


def test_non_equi_outer_joins(s):
    s2 = sail_amd.SessionContext(device="cpu")
    s2.create_dataframe({"a": [1, 5], "x": ["p", "q"]}, name="nl")
    s2.create_dataframe({"b": [2, 9], "y": ["r", "t"]}, name="nr")
    assert s2.sql("SELECT * FROM nl FULL JOIN nr ON a > b "
                  "ORDER BY a NULLS LAST, b NULLS LAST").collect() == [
        (1, "p", None, None), (5, "q", 2, "r"), (None, None, 9, "t")]
    assert s2.sql("SELECT * FROM nl RIGHT JOIN nr ON a > b ORDER BY b").collect() == [
        (5, "q", 2, "r"), (None, None, 9, "t")]
    assert s2.sql("SELECT a FROM nl LEFT SEMI JOIN nr ON a > b").collect() == [(5,)]
    assert s2.sql("SELECT a FROM nl LEFT ANTI JOIN nr ON a > b").collect() == [(1,)]


def test_todo_closures(s):
    s2 = sail_amd.SessionContext(device="cpu")
    # raw-string (non-dictionary) min/max
    names = [f"name_{i:04d}" for i in range(80)]
    import random

    random.seed(1)
    random.shuffle(names)
    s2.create_dataframe({"g": [i % 2 for i in range(80)], "c": names}, name="rm")
    rows = s2.sql("SELECT g, min(c), max(c) FROM rm GROUP BY g ORDER BY g").collect()
    assert rows == [(0, "name_0000", "name_0079"), (1, "name_0001", "name_0077")]
    # unix_timestamp on strings
    assert s2.sql("SELECT unix_timestamp('2024-01-01 00:00:00')").collect() == [
        (1704067200,)]
    # IN with non-literal (column-dependent) values, incl. 3VL nulls
    s2.create_dataframe({"a": [1, 2, 3], "x": [1, 5, None]}, name="nv")
    rows = s2.sql("SELECT a, a IN (x - 1, x) FROM nv ORDER BY a").collect()
    assert rows == [(1, True), (2, False), (3, None)]


def test_values_table_factor_and_expressions(s):
    # VALUES as a FROM-clause table factor with column aliases
    assert s.sql("SELECT * FROM VALUES (1+1, upper('a')), (10, 'b') AS t(x, y)"
                 ).collect() == [(2, "A"), (10, "b")]
    # expression rows route through one-row SELECT union
    assert s.sql("SELECT x*2 FROM VALUES (abs(-5)) t(x)").collect() == [(10,)]
    # literal rows keep the LocalRelation fast path
    assert s.sql("VALUES (1, 2), (3, 4)").collect() == [(1, 2), (3, 4)]
    assert s.sql("SELECT y, sum(x) FROM VALUES (1,'a'),(2,'a'),(3,'b') AS t(x,y) "
                 "GROUP BY y ORDER BY y").collect() == [("a", 3), ("b", 3)]


def test_url_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT parse_url('https://u:p@spark.apache.org:8080/path?query=1#Ref', 'HOST')") \
        == [("spark.apache.org",)]
    assert q("SELECT parse_url('https://h/p?a=1&b=2', 'QUERY', 'b')") == [("2",)]
    assert q("SELECT parse_url('https://u:p@h/p', 'USERINFO'), "
             "parse_url('https://h/p?a=1', 'FILE'), "
             "parse_url('https://h/p', 'PROTOCOL')") == [("u:p", "/p?a=1", "https")]
    assert q("SELECT url_encode('hello world/x'), url_decode('hello+world%2Fx')") \
        == [("hello+world%2Fx", "hello world/x")]


def test_xpath_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT xpath('<a><b>b1</b><b>b2</b></a>', 'a/b/text()')") \
        == [(["b1", "b2"],)]
    assert q("SELECT xpath_string('<a><b>bb</b></a>', 'a/b'), "
             "xpath_int('<a><b>3</b></a>', 'a/b'), "
             "xpath_double('<a><b>2.5</b></a>', 'a/b'), "
             "xpath_boolean('<a><b>1</b></a>', 'a/c')") == [("bb", 3, 2.5, False)]
    assert q("SELECT xpath('<r><x id=\"7\"/><x id=\"9\"/></r>', 'r/x/@id')") \
        == [(["7", "9"],)]
    assert q("SELECT xpath('<a><b><c>1</c></b><b><c>2</c></b></a>', '//c/text()')") \
        == [(["1", "2"],)]
    # malformed XML -> null
    assert q("SELECT xpath_string('<oops', 'a')") == [(None,)]


def test_csv_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT from_csv('1,apple', 'a INT, b STRING').a, "
             "from_csv('1,apple', 'a INT, b STRING').b") == [(1, "apple")]
    assert q("SELECT to_csv(named_struct('a', 1, 'b', 'x'))") == [("1,x",)]
    assert q("SELECT schema_of_csv('1,abc,2.5')") \
        == [("STRUCT<_c0: BIGINT, _c1: STRING, _c2: DOUBLE>",)]


def test_variant_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("""SELECT parse_json('{"b": 1, "a": [1,2]}')""") \
        == [('{"b":1,"a":[1,2]}',)]
    assert q("SELECT try_parse_json('oops')") == [(None,)]
    assert q("""SELECT variant_get(parse_json('{"a":[1,{"b":5}]}'), '$.a[1].b')""") \
        == [("5",)]
    assert q("SELECT is_variant_null(parse_json('null'))") == [(True,)]
    assert q("""SELECT schema_of_variant('{"x": 1.5, "y": "s"}')""") \
        == [("OBJECT<x: DOUBLE, y: STRING>",)]
    with pytest.raises(Exception):
        q("SELECT parse_json('bad')")


def test_json_misc_and_checks(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT json_array_length('[1,2,3]'), json_array_length('{}')") \
        == [(3, None)]
    assert q("""SELECT json_object_keys('{"a":1,"b":2}')""") == [(["a", "b"],)]
    assert q("SELECT luhn_check('79927398713'), luhn_check('79927398714')") \
        == [(True, False)]
    assert q("SELECT crc32c('abc')") == [(910901175,)]


def test_collection_builders(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT arrays_zip(xs, ys) FROM (SELECT array(1,2) AS xs, "
             "array('a','b') AS ys)") == \
        [([{"xs": 1, "ys": "a"}, {"xs": 2, "ys": "b"}],)]
    assert q("SELECT array_insert(array(1,3), 2, 2), array_insert(array(1,2), -1, 9)") \
        == [([1, 2, 3], [1, 2, 9])]
    assert q("SELECT array_contains_all(array(1,2,3), array(1,2)), "
             "array_contains_all(array(1), array(2))") == [(True, False)]
    assert q("SELECT array_concat(array(1), array(2,3)), concat(array(4), array(5))") \
        == [([1, 2, 3], [4, 5])]
    assert q("SELECT map_concat(map('a',1), map('b',2))") == [({"a": 1, "b": 2},)]
    assert q("SELECT map_entries(map('a',1))") == [([{"key": "a", "value": 1}],)]
    assert q("SELECT map_from_entries(map_entries(map('a',1,'b',2)))") \
        == [({"a": 1, "b": 2},)]
    assert q("SELECT str_to_map('a:1,b:2')") == [({"a": "1", "b": "2"},)]
    assert q("SELECT map_zip_with(map('a',1,'b',2), map('a',10,'c',30), "
             "(k,v1,v2) -> coalesce(v1,0) + coalesce(v2,0))") \
        == [({"a": 11, "b": 2, "c": 30},)]


def test_vector_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT vector_norm(array(3.0,4.0)), "
             "vector_inner_product(array(1.0,2.0), array(3.0,4.0)), "
             "vector_l2_distance(array(0.0,0.0), array(3.0,4.0))") \
        == [(5.0, 11.0, 5.0)]
    (vec,), = q("SELECT vector_normalize(array(3.0,4.0))")
    assert vec == [0.6, 0.8]


def test_null_safe_and_try_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT 1 <=> NULL, NULL <=> NULL, 2 <=> 2") == [(False, True, True)]
    assert q("SELECT equal_null(1, NULL), equal_null(NULL, NULL)") == [(False, True)]
    assert q("SELECT nullifzero(0), nullifzero(5), zeroifnull(NULL), zeroifnull(7)") \
        == [(None, 5, 0, 7)]
    assert q("SELECT try_mod(5, 0), try_mod(7, 3)") == [(None, 1)]


def test_misc_string_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT strpos('hello', 'l')") == [(3,)]
    assert q("SELECT quote(\"Don't\")") == [("'Don\\'t'",)]
    assert q("SELECT mask('AbCD123-@$#'), mask('AbCD123-@$#', 'Q', 'q', 'd', 'o')") \
        == [("XxXXnnn-@$#", "QqQQdddoooo")]
    assert q("SELECT regexp_count('Steven', 'e'), regexp_instr('hello world', 'o'), "
             "regexp_substr('hello world', 'o.')") == [(2, 5, "o ")]
    assert q("SELECT to_number('$1,234.56', '999'), try_to_number('oops', '999')") \
        == [(1234.56, None)]
    assert q("SELECT to_binary('616263', 'hex'), try_to_binary('_bad_', 'base64')") \
        == [(b"abc", None)]  # BINARY surfaces as bytes (Spark parity)
    assert q("SELECT to_varchar(123), to_char(1.5, '9.9')") == [("123", "1.5")]


def test_timestamp_arithmetic_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT timestampdiff(HOUR, TIMESTAMP '2024-01-01 00:00:00', "
             "TIMESTAMP '2024-01-01 05:30:00')") == [(5,)]
    # month-add clamps to end of shorter month (Java Calendar semantics)
    assert q("SELECT timestampadd(MONTH, 1, TIMESTAMP '2024-01-31 00:00:00')") \
        == [(1709164800000000,)]  # 2024-02-29
    # partial months do not count
    assert q("SELECT timestampdiff(MONTH, TIMESTAMP '2024-01-31 00:00:00', "
             "TIMESTAMP '2024-02-29 00:00:00')") == [(0,)]
    assert q("SELECT convert_timezone('UTC', 'America/Los_Angeles', "
             "TIMESTAMP '2024-01-01 08:00:00')") == [(1704067200000000,)]
    assert q("SELECT dayname(DATE '2024-01-01'), date_from_unix_date(1)") \
        == [("Mon", __import__("datetime").date(1970, 1, 2))]


def test_bit_and_random_functions(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT getbit(5, 0), getbit(5, 1), bit_get(5, 2)") == [(1, 0, 1)]
    assert q("SELECT random() BETWEEN 0 AND 1, uniform(0, 10) BETWEEN 0 AND 10, "
             "length(randstr(8))") == [(True, True, 8)]
    # seeded variants are deterministic
    assert q("SELECT uniform(0, 100, 42)") == q("SELECT uniform(0, 100, 42)")
    assert q("SELECT randstr(6, 1)") == q("SELECT randstr(6, 1)")


def test_ordered_set_and_vector_aggregates(s):
    s.create_dataframe({"g": [1, 1, 1, 1, 2, 2], "x": [10, 20, 30, 40, 5, 15]},
                       name="osa")
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT g, percentile_cont(0.5) WITHIN GROUP (ORDER BY x) "
             "FROM osa GROUP BY g ORDER BY g") == [(1, 25.0), (2, 10.0)]
    assert q("SELECT g, percentile_disc(0.5) WITHIN GROUP (ORDER BY x) "
             "FROM osa GROUP BY g ORDER BY g") == [(1, 20), (2, 5)]
    assert q("SELECT g, percentile_disc(0.5) WITHIN GROUP (ORDER BY x DESC) "
             "FROM osa GROUP BY g ORDER BY g") == [(1, 30), (2, 15)]
    assert q("SELECT std(x) = stddev(x) FROM osa") == [(True,)]
    assert q("SELECT g, vector_sum(v), vector_avg(v) FROM "
             "(SELECT g, array(1.0*g, 2.0) AS v FROM osa) GROUP BY g ORDER BY g") \
        == [(1, [4.0, 8.0], [1.0, 2.0]), (2, [4.0, 4.0], [2.0, 2.0])]
    (hist,), = q("SELECT histogram_numeric(x, 2) FROM osa WHERE g = 1")
    assert len(hist) == 2 and sum(b["y"] for b in hist) == 4.0


def test_xml_struct_and_collation(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT from_xml('<r><a>1</a><b>x</b></r>', 'a INT, b STRING').a, "
             "from_xml('<r><a>1</a><b>x</b></r>', 'a INT, b STRING').b") \
        == [(1, "x")]
    assert q("SELECT to_xml(named_struct('a', 1, 'b', 'x'))") \
        == [("<ROW><a>1</a><b>x</b></ROW>",)]
    assert q("SELECT schema_of_xml('<r><a>1</a><b>s</b></r>')") \
        == [("STRUCT<a: BIGINT, b: STRING>",)]
    assert q("SELECT collate('x', 'UTF8_BINARY'), collation('y')") \
        == [("x", "UTF8_BINARY")]


def test_interval_makers(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT TIMESTAMP '2024-01-01 00:00:00' + make_dt_interval(1, 2, 3, 4.5)") \
        == [(1704160984500000,)]  # +1d 2h 3m 4.5s
    assert q("SELECT DATE '2024-01-15' + make_ym_interval(1, 1)") \
        == [(__import__("datetime").date(2025, 2, 15),)]


def test_trivial_parity_batch(s):
    q = lambda x: s.sql(x).collect()  # noqa: E731
    assert q("SELECT round(cot(1.0), 4), round(sec(1.0), 4)") == [(0.6421, 1.8508)]
    assert q("SELECT current_timezone(), is_valid_utf8('x')") == [("UTC", True)]
    assert q("""SELECT variant_to_json(parse_json('{"a":1}')), """
             "is_valid_variant('{}'), is_valid_variant('no')") \
        == [('{"a":1}', True, False)]
    assert q("SELECT try_url_decode('a%20b')") == [("a b",)]
    assert q("SELECT bitmap_bit_position(32769), bitmap_bucket_number(32769)") \
        == [(0, 2)]
    assert q("SELECT DATE '2020-01-01' + make_interval(1, 0, 1, 1)") \
        == [(__import__("datetime").date(2021, 1, 9),)]
    # month interval on TIMESTAMP clamps to month end
    assert q("SELECT TIMESTAMP '2020-01-31 00:00:00' + INTERVAL 1 MONTH") \
        == [(1582934400000000,)]  # 2020-02-29


def test_order_by_qualified_group_key(s):
    """ORDER BY t.col where col is a grouped output key: qualifiers don't
    survive aggregation, but Spark accepts the qualified form."""
    s.sql("CREATE TEMP VIEW oq_t AS SELECT * FROM VALUES "
          "(1, 'b'), (2, 'a'), (3, 'a') AS t(v, g)")
    r = s.sql("SELECT t.g, sum(t.v) FROM oq_t t GROUP BY t.g ORDER BY t.g"
              ).collect()
    assert r == [("a", 5), ("b", 1)]


def test_null_semantics_and_intervals(s):
    q = s.sql
    # IN with NULL list entries: SQL three-valued logic
    assert q("SELECT 1 IN (2, NULL), 1 IN (1, NULL), "
             "1 NOT IN (2, NULL), 1 NOT IN (1, NULL)").collect() == \
        [(None, True, None, False)]
    # least/greatest skip nulls; all-null -> null
    assert q("SELECT least(1, NULL, 3), greatest(NULL, NULL), "
             "greatest(1, NULL, 5)").collect() == [(1, None, 5)]
    # multi-unit interval literals
    assert q("SELECT timestamp '2024-01-01 00:00:00' + "
             "INTERVAL '1 02:03:04.5' DAY TO SECOND").collect() == \
        [(1704160984500000,)]
    assert q("SELECT date '2024-01-01' + INTERVAL '1-2' YEAR TO MONTH"
             ).collect()[0][0].isoformat() == "2025-03-01"
    assert q("SELECT INTERVAL '02:30' HOUR TO MINUTE").collect() == \
        [("INTERVAL '2 hours 30 minutes'",)]
