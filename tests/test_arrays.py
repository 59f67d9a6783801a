"""Array type, functions, collect aggregates, explode generators
(ref: sail-plan function/scalar collection + generator.rs semantics)."""
import pytest

import sail_amd
from sail_amd.engine import types as T


@pytest.fixture()
def s():
    return sail_amd.SessionContext(device="cpu")


def test_array_construction_and_size(s):
    assert s.sql("SELECT array(1, 2, 3)").collect() == [([1, 2, 3],)]
    assert s.sql("SELECT size(array(1,2,3)), cardinality(array())").collect()[0][0] == 3


def test_element_at(s):
    rows = s.sql("SELECT element_at(array(10,20,30), 2), element_at(array(10,20), -1), "
                 "element_at(array(10), 5)").collect()
    assert rows == [(20, 20, None)]


def test_array_predicates(s):
    rows = s.sql("SELECT array_contains(array(1,2),2), array_contains(array(1),9), "
                 "array_position(array(5,6,7),7), array_position(array(5),9)").collect()
    assert rows == [(True, False, 3, 0)]


def test_array_minmax_sort_distinct(s):
    rows = s.sql("SELECT array_min(array(3,1,2)), array_max(array(3,9)), "
                 "sort_array(array(3,1,2)), sort_array(array(1,3), false), "
                 "array_distinct(array(1,2,1,3,2))").collect()
    assert rows == [(1, 9, [1, 2, 3], [3, 1], [1, 2, 3])]


def test_array_join_slice_sequence(s):
    rows = s.sql("SELECT array_join(array('a','b','c'), '-'), "
                 "slice(array(1,2,3,4), 2, 2), slice(array(1,2,3,4), -2, 9), "
                 "sequence(1, 4), sequence(4, 1)").collect()
    assert rows == [("a-b-c", [2, 3], [3, 4], [1, 2, 3, 4], [4, 3, 2, 1])]


def test_split_and_overlap(s):
    rows = s.sql("SELECT split('a,b,c', ','), arrays_overlap(array(1,2), array(2,9)), "
                 "arrays_overlap(array(1), array(9))").collect()
    assert rows == [(["a", "b", "c"], True, False)]


def test_array_column_ops(s):
    s.create_dataframe({"v": [1, 2, 3]}, schema={"v": T.I64}, name="t")
    rows = s.sql("SELECT v, size(sequence(1, v)) AS n FROM t ORDER BY v").collect()
    assert rows == [(1, 1), (2, 2), (3, 3)]


def test_collect_list_set(s):
    s.create_dataframe({"g": ["a", "a", "b", "a"], "v": [1, 2, 3, 2]}, name="t")
    rows = s.sql("SELECT g, collect_list(v), collect_set(v) FROM t GROUP BY g ORDER BY g").collect()
    assert rows[0][0] == "a" and rows[0][1] == [1, 2, 2] and sorted(rows[0][2]) == [1, 2]
    assert rows[1] == ("b", [3], [3])


def test_collect_list_strings(s):
    s.create_dataframe({"g": [1, 1, 2], "c": ["x", "y", "z"]}, name="t")
    rows = s.sql("SELECT g, collect_list(c) FROM t GROUP BY g ORDER BY g").collect()
    assert rows == [(1, ["x", "y"]), (2, ["z"])]


def test_explode(s):
    s.create_dataframe({"k": ["a", "b"], "v": [2, 3]}, name="t")
    rows = s.sql("SELECT k, explode(sequence(1, v)) AS e FROM t ORDER BY k, e").collect()
    assert rows == [("a", 1), ("a", 2), ("b", 1), ("b", 2), ("b", 3)]


def test_posexplode(s):
    rows = s.sql("SELECT posexplode(array(7, 8))").collect()
    assert rows == [(0, 7), (1, 8)]


def test_explode_outer_empty_and_null(s):
    s.create_dataframe({"k": ["a", "b"], "v": [2, 0]}, name="t")
    rows = s.sql("SELECT k, explode_outer(slice(sequence(1,3), 1, v)) AS e "
                 "FROM t ORDER BY k, e NULLS LAST").collect()
    assert rows == [("a", 1), ("a", 2), ("b", None)]
    rows = s.sql("SELECT k, explode(slice(sequence(1,3), 1, v)) AS e "
                 "FROM t ORDER BY k, e").collect()
    assert rows == [("a", 1), ("a", 2)]  # inner explode drops empty


def test_explode_then_aggregate(s):
    s.create_dataframe({"k": ["a", "b"], "v": [3, 4]}, name="t")
    rows = s.sql("WITH e AS (SELECT k, explode(sequence(1, v)) AS x FROM t) "
                 "SELECT k, sum(x) FROM e GROUP BY k ORDER BY k").collect()
    assert rows == [("a", 6), ("b", 10)]


def test_array_to_arrow_roundtrip(s):
    s.create_dataframe({"g": ["a", "a", "b"], "v": [1, 2, 3]}, name="t")
    t = s.sql("SELECT g, collect_list(v) AS vs FROM t GROUP BY g ORDER BY g").to_arrow()
    assert t.column("vs").to_pylist() == [[1, 2], [3]]


def test_split_part_still_scalar(s):
    assert s.sql("SELECT split_part('a:b:c', ':', 2)").collect() == [("b",)]


def test_lateral_view_explode(s):
    s.create_dataframe({"k": ["a", "b"], "v": [2, 3]}, name="lt")
    rows = s.sql("SELECT k, e FROM lt LATERAL VIEW explode(sequence(1, v)) ex AS e "
                 "ORDER BY k, e").collect()
    assert rows == [("a", 1), ("a", 2), ("b", 1), ("b", 2), ("b", 3)]


def test_lateral_view_posexplode_qualified(s):
    s.create_dataframe({"k": ["a"], "v": [2]}, name="lt2")
    rows = s.sql("SELECT k, ex.p, ex.c FROM lt2 "
                 "LATERAL VIEW posexplode(sequence(1, v)) ex AS p, c ORDER BY p").collect()
    assert rows == [("a", 0, 1), ("a", 1, 2)]


def test_lateral_view_outer(s):
    s.create_dataframe({"k": ["a", "b"], "v": [1, 0]}, name="lt3")
    rows = s.sql("SELECT k, e FROM lt3 "
                 "LATERAL VIEW OUTER explode(slice(sequence(1,3), 1, v)) ex AS e "
                 "ORDER BY k, e NULLS LAST").collect()
    assert rows == [("a", 1), ("b", None)]


def test_lateral_view_then_join(s):
    s.create_dataframe({"k": ["a", "b"], "v": [2, 1]}, name="lt4")
    s.create_dataframe({"e": [1, 2], "w": [10, 20]}, name="lt5")
    rows = s.sql("SELECT k, e, w FROM lt4 LATERAL VIEW explode(sequence(1, v)) x AS e "
                 "JOIN lt5 USING (e) ORDER BY k, e").collect()
    assert rows == [("a", 1, 10), ("a", 2, 20), ("b", 1, 10)]


def test_lambda_transform(s):
    assert s.sql("SELECT transform(array(1,2,3), x -> x * 10)").collect() == [([10, 20, 30],)]
    assert s.sql("SELECT transform(array(1,2,3), (x, i) -> x + i)").collect() == [([1, 3, 5],)]


def test_lambda_filter_exists_forall(s):
    rows = s.sql("SELECT filter(array(1,2,3,4), x -> x % 2 = 0), "
                 "exists(array(1,2), x -> x > 1), forall(array(1,2), x -> x > 1), "
                 "forall(array(), x -> x > 1)").collect()
    assert rows == [([2, 4], True, False, True)]


def test_lambda_captures_outer_column(s):
    s.create_dataframe({"base": [10, 20], "v": [2, 3]}, name="hof")
    rows = s.sql("SELECT transform(sequence(1, v), x -> x + base) FROM hof ORDER BY base").collect()
    assert rows == [([11, 12],), ([21, 22, 23],)]
    rows = s.sql("SELECT filter(sequence(1, 5), x -> x <= v) FROM hof ORDER BY base").collect()
    assert rows == [([1, 2],), ([1, 2, 3],)]


def test_lambda_string_elements(s):
    rows = s.sql("SELECT transform(split('a,bb,ccc', ','), x -> length(x)), "
                 "filter(split('a,bb,ccc', ','), x -> length(x) > 1)").collect()
    assert rows == [([1, 2, 3], ["bb", "ccc"])]


def test_zip_with(s):
    rows = s.sql("SELECT zip_with(array(1,2,3), array(10,20,30), (x, y) -> x + y), "
                 "zip_with(array(1,2), array(10,20,30), (x, y) -> y)").collect()
    assert rows == [([11, 22, 33], [10, 20, 30])]


def test_aggregate_reduce(s):
    rows = s.sql("SELECT aggregate(array(1,2,3,4), 0, (acc, x) -> acc + x), "
                 "aggregate(array(1,2,3), 1, (acc, x) -> acc * x, acc -> acc * 10)").collect()
    assert rows == [(10, 60)]
    s.create_dataframe({"v": [2, 4]}, name="agg_t")
    rows = s.sql("SELECT aggregate(sequence(1, v), 0, (acc, x) -> acc + x) "
                 "FROM agg_t ORDER BY v").collect()
    assert rows == [(3,), (10,)]


def test_lambda_pushdown_through_projection(s):
    """Filter with a lambda pushed through a subquery projection must keep
    param indices intact (regression: substitute_refs corrupted them)."""
    s.create_dataframe({"v": [2, 5]}, name="lpd")
    q = ("SELECT v FROM (SELECT sequence(1, v) AS a, v FROM lpd) x "
         "WHERE exists(a, e -> e = v) ORDER BY v")
    assert s.sql(q).collect() == [(2,), (5,)]
    q2 = ("SELECT v FROM (SELECT sequence(1, v) AS a, v FROM lpd) x "
          "WHERE size(filter(a, e -> e > 1)) > 3")
    assert s.sql(q2).collect() == [(5,)]


def test_array_set_ops(s):
    rows = s.sql("SELECT array_union(array(1,2), array(2,3)), "
                 "array_intersect(array(1,2), array(2,3)), "
                 "array_except(array(1,2), array(2,3))").collect()
    assert rows == [([1, 2, 3], [2], [1])]
    rows = s.sql("SELECT array_remove(array(1,2,1), 1), "
                 "array_compact(array(1, NULL, 2)), "
                 "flatten(array(array(1,2), array(3))), array_repeat('x', 3), "
                 "array_append(array(1), 2), array_prepend(array(1), 0)").collect()
    assert rows == [([2], [1, 2], [1, 2, 3], ["x", "x", "x"], [1, 2], [0, 1])]


def test_map_lambdas(s):
    rows = s.sql("SELECT transform_values(map('a',1,'b',2), (k,v) -> v * 10), "
                 "transform_keys(map('a',1), (k, v) -> upper(k)), "
                 "map_filter(map('a',1,'b',2), (k,v) -> v > 1)").collect()
    assert rows == [({"a": 10, "b": 20}, {"A": 1}, {"b": 2})]


def test_datetime_formatting(s):
    rows = s.sql("SELECT date_format(DATE '2024-03-05', 'yyyy-MM-dd'), "
                 "from_unixtime(86400), "
                 "datepart('year', DATE '2024-03-05'), "
                 "unix_timestamp(to_timestamp('1970-01-02 00:00:00'))").collect()
    assert rows == [("2024-03-05", "1970-01-02 00:00:00", 2024, 86400)]


def test_generator_breadth(s):
    """explode(map) -> key/value, inline(array<struct>), stack(n, ...)
    (ref: sail-plan function/generator.rs)."""
    q = s.sql
    assert q("SELECT explode(map('a', 1, 'b', 2))").collect() == \
        [("a", 1), ("b", 2)]
    assert q("SELECT posexplode(map('a', 1))").collect() == [(0, "a", 1)]
    assert q("SELECT inline(array(named_struct('a', 1, 'b', 'x'), "
             "named_struct('a', 2, 'b', 'y')))").collect() == \
        [(1, "x"), (2, "y")]
    assert q("SELECT stack(2, 1, 'a', 2, 'b')").collect() == \
        [(1, "a"), (2, "b")]
    assert q("SELECT stack(3, 1, 2, 3, 4, 5)").collect() == \
        [(1, 2), (3, 4), (5, None)]
    s.sql("CREATE TEMP VIEW gen_g AS SELECT * FROM VALUES (1), (2) AS "
          "t(id)")
    assert q("SELECT id, explode(map('k', id)) FROM gen_g").collect() == \
        [(1, "k", 1), (2, "k", 2)]


def test_nested_element_field_access(s):
    assert s.sql("SELECT array(named_struct('a', 1, 'b', 'x'))[0].a"
                 ).collect() == [(1,)]
    assert s.sql("SELECT element_at(array(array(1,2), array(3)), 2)"
                 ).collect() == [([3],)]


def test_if_function_types(s):
    s.sql("CREATE TEMP VIEW if_t AS SELECT * FROM VALUES (1), (2) AS t(v)")
    assert s.sql("SELECT v, IF(v > 1, 'big', 'small') FROM if_t"
                 ).collect() == [(1, "small"), (2, "big")]
    assert s.sql("SELECT IF(false, 1.5, 2.5)").collect() == [(2.5,)]
    assert s.sql("SELECT try_cast('abc' AS INT), try_cast('12' AS INT)"
                 ).collect() == [(None, 12)]


def test_array_sort_comparator(s):
    q = s.sql
    assert q("SELECT array_sort(array(3,1,2), (a,b) -> b - a)"
             ).collect() == [([3, 2, 1],)]
    assert q("SELECT array_sort(array(3,1,2), (a,b) -> a - b)"
             ).collect() == [([1, 2, 3],)]
    assert q("SELECT array_sort(array('bb','a','ccc'), "
             "(a,b) -> length(a) - length(b))").collect() == \
        [(["a", "bb", "ccc"],)]
    s.sql("CREATE TEMP VIEW asrt2 AS SELECT * FROM VALUES "
          "(array(5,2,9)), (array(1)) AS t(a)")
    assert q("SELECT array_sort(a, (x,y) -> y - x) FROM asrt2"
             ).collect() == [([9, 5, 2],), ([1],)]
