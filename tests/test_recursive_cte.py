"""WITH RECURSIVE (ref: sail-plan resolver/query/recursion.rs role)."""
import pytest

import sail_amd


@pytest.fixture()
def s():
    return sail_amd.SessionContext(device="cpu")


def test_numbers(s):
    rows = s.sql("WITH RECURSIVE n(x) AS (SELECT 1 UNION ALL "
                 "SELECT x + 1 FROM n WHERE x < 10) "
                 "SELECT sum(x), count(*) FROM n").collect()
    assert rows == [(55, 10)]


def test_cyclic_graph_union_distinct_terminates(s):
    s.create_dataframe({"src": [1, 2, 3, 3], "dst": [2, 3, 1, 4]}, name="edges")
    rows = s.sql(
        "WITH RECURSIVE reach(node) AS ("
        " SELECT 1 AS node"
        " UNION"
        " SELECT e.dst FROM reach r JOIN edges e ON e.src = r.node"
        ") SELECT node FROM reach ORDER BY node").collect()
    assert rows == [(1,), (2,), (3,), (4,)]


def test_hierarchy_with_strings(s):
    s.create_dataframe({"id": [1, 2, 3, 4], "mgr": [None, 1, 1, 2],
                        "nm": ["ceo", "a", "b", "c"]}, name="emp")
    rows = s.sql(
        "WITH RECURSIVE chain(id, nm, depth) AS ("
        " SELECT id, nm, 0 FROM emp WHERE mgr IS NULL"
        " UNION ALL"
        " SELECT e.id, e.nm, c.depth + 1 FROM emp e JOIN chain c ON e.mgr = c.id"
        ") SELECT nm, depth FROM chain ORDER BY depth, nm").collect()
    assert rows == [("ceo", 0), ("a", 1), ("b", 1), ("c", 2)]


def test_iteration_limit(s):
    with pytest.raises(Exception, match="exceeded"):
        s.sql("WITH RECURSIVE n(x) AS (SELECT 1 UNION ALL SELECT x + 1 FROM n) "
              "SELECT count(*) FROM n").collect()
    # raising the limit makes it fail later, proving the conf is honored
    s.conf["sail.execution.max_recursion"] = "5"
    with pytest.raises(Exception, match="exceeded 5"):
        s.sql("WITH RECURSIVE n(x) AS (SELECT 1 UNION ALL SELECT x + 1 FROM n "
              "WHERE x < 50) SELECT count(*) FROM n").collect()


def test_mixed_recursive_and_plain_ctes(s):
    rows = s.sql(
        "WITH RECURSIVE base AS (SELECT 3 AS lim), "
        "n(x) AS (SELECT 1 UNION ALL SELECT x + 1 FROM n WHERE x < (SELECT lim FROM base)) "
        "SELECT count(*) FROM n").collect()
    assert rows == [(3,)]
