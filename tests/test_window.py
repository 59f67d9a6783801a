"""Window function coverage (ref: sail-function/src/window, BoundedWindowAggExec role)."""
import pytest

import sail_amd
from sail_amd.engine import types as T


@pytest.fixture()
def s():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe(
        {"g": ["a", "a", "a", "b", "b"], "o": [1, 2, 3, 1, 2],
         "v": [10, 20, 30, 5, 15]},
        schema={"g": T.STRING, "o": T.I32, "v": T.I64}, name="w")
    return s


def test_row_number_rank(s):
    rows = s.sql("SELECT g, o, row_number() OVER (PARTITION BY g ORDER BY o DESC) AS rn "
                 "FROM w ORDER BY g, o").collect()
    assert rows == [("a", 1, 3), ("a", 2, 2), ("a", 3, 1), ("b", 1, 2), ("b", 2, 1)]


def test_rank_with_ties(s):
    s.create_dataframe({"x": [1, 1, 2, 3, 3, 3]}, name="r")
    rows = s.sql("SELECT x, rank() OVER (ORDER BY x) AS r, dense_rank() OVER (ORDER BY x) AS d "
                 "FROM r ORDER BY x").collect()
    assert rows == [(1, 1, 1), (1, 1, 1), (2, 3, 2), (3, 4, 3), (3, 4, 3), (3, 4, 3)]


def test_running_sum(s):
    rows = s.sql("SELECT g, o, sum(v) OVER (PARTITION BY g ORDER BY o) AS rs "
                 "FROM w ORDER BY g, o").collect()
    assert rows == [("a", 1, 10), ("a", 2, 30), ("a", 3, 60), ("b", 1, 5), ("b", 2, 20)]


def test_whole_partition_agg(s):
    rows = s.sql("SELECT g, v, sum(v) OVER (PARTITION BY g) AS tot FROM w ORDER BY g, o").collect()
    assert rows == [("a", 10, 60), ("a", 20, 60), ("a", 30, 60), ("b", 5, 20), ("b", 15, 20)]


def test_lag_lead(s):
    rows = s.sql("SELECT g, o, lag(v) OVER (PARTITION BY g ORDER BY o) AS lg, "
                 "lead(v) OVER (PARTITION BY g ORDER BY o) AS ld FROM w ORDER BY g, o").collect()
    assert rows == [("a", 1, None, 20), ("a", 2, 10, 30), ("a", 3, 20, None),
                    ("b", 1, None, 15), ("b", 2, 5, None)]


def test_ntile(s):
    rows = s.sql("SELECT o, ntile(2) OVER (ORDER BY o) AS nt FROM w WHERE g = 'a' ORDER BY o").collect()
    assert rows == [(1, 1), (2, 1), (3, 2)]


def test_bounded_rows_frames(s):
    s.create_dataframe({"o": [1, 2, 3, 4], "v": [10, 20, 30, 40]}, name="bf")
    rows = s.sql("SELECT o, sum(v) OVER (ORDER BY o ROWS BETWEEN 1 PRECEDING AND 1 FOLLOWING) "
                 "FROM bf ORDER BY o").collect()
    assert rows == [(1, 30), (2, 60), (3, 90), (4, 70)]
    rows = s.sql("SELECT o, avg(v) OVER (ORDER BY o ROWS BETWEEN 2 PRECEDING AND CURRENT ROW) "
                 "FROM bf ORDER BY o").collect()
    assert rows == [(1, 10.0), (2, 15.0), (3, 20.0), (4, 30.0)]
    # empty trailing frames -> count 0
    rows = s.sql("SELECT o, count(*) OVER (ORDER BY o ROWS BETWEEN 1 FOLLOWING AND 2 FOLLOWING) "
                 "FROM bf ORDER BY o").collect()
    assert rows == [(1, 2), (2, 2), (3, 1), (4, 0)]


def test_bounded_frames_with_partitions(s):
    rows = s.sql("SELECT g, o, sum(v) OVER (PARTITION BY g ORDER BY o "
                 "ROWS BETWEEN 1 PRECEDING AND CURRENT ROW) FROM w ORDER BY g, o").collect()
    assert rows == [("a", 1, 10), ("a", 2, 30), ("a", 3, 50), ("b", 1, 5), ("b", 2, 20)]


def test_first_last_nth_cume(s):
    s.create_dataframe({"o": [1, 2, 3, 4], "v": [10, 20, 30, 40]}, name="fl")
    rows = s.sql("SELECT o, first_value(v) OVER (ORDER BY o DESC), "
                 "last_value(v) OVER (ORDER BY o), "
                 "nth_value(v, 2) OVER (ORDER BY o), "
                 "cume_dist() OVER (ORDER BY o) FROM fl ORDER BY o").collect()
    assert rows == [(1, 40, 10, None, 0.25), (2, 40, 20, 20, 0.5),
                    (3, 40, 30, 20, 0.75), (4, 40, 40, 20, 1.0)]


def test_lag_lead_default_values(s):
    s.sql("CREATE TEMP VIEW wd_t AS SELECT * FROM VALUES "
          "(1, 10), (2, NULL), (3, 30) AS t(o, x)")
    q = s.sql
    assert q("SELECT o, lag(x, 1, -5) OVER (ORDER BY o) FROM wd_t"
             ).collect() == [(1, -5), (2, 10), (3, None)]
    assert q("SELECT o, lead(o, 1, 99) OVER (ORDER BY o) FROM wd_t"
             ).collect() == [(1, 2), (2, 3), (3, 99)]
    assert q("SELECT o, lag(CAST(o AS STRING), 1, 'none') "
             "OVER (ORDER BY o) FROM wd_t").collect() == \
        [(1, "none"), (2, "1"), (3, "2")]


def test_range_frames(s):
    s.sql("CREATE TEMP VIEW rf_t AS SELECT * FROM VALUES "
          "(1,'a'),(2,'a'),(4,'a'),(5,'b'),(7,'b') AS t(v, k)")
    q = s.sql
    assert q("SELECT v, sum(v) OVER (PARTITION BY k ORDER BY v "
             "RANGE BETWEEN 1 PRECEDING AND CURRENT ROW) FROM rf_t "
             "ORDER BY k, v").collect() == \
        [(1, 1), (2, 3), (4, 4), (5, 5), (7, 7)]
    assert q("SELECT v, sum(v) OVER (PARTITION BY k ORDER BY v "
             "RANGE BETWEEN CURRENT ROW AND 2 FOLLOWING) FROM rf_t "
             "ORDER BY k, v").collect() == \
        [(1, 3), (2, 6), (4, 4), (5, 12), (7, 7)]
    # duplicate order values are peers: they share the frame
    s.sql("CREATE TEMP VIEW rf_t2 AS SELECT * FROM VALUES "
          "(1),(2),(2),(3) AS t(v)")
    assert q("SELECT v, count(*) OVER (ORDER BY v RANGE BETWEEN "
             "CURRENT ROW AND CURRENT ROW) FROM rf_t2").collect() == \
        [(1, 1), (2, 2), (2, 2), (3, 1)]
    assert q("SELECT v, avg(v) OVER (ORDER BY v RANGE BETWEEN "
             "UNBOUNDED PRECEDING AND 1 FOLLOWING) FROM rf_t2"
             ).collect() == [(1, pytest.approx(5 / 3)), (2, 2.0),
                             (2, 2.0), (3, 2.0)]
