"""Gold plan-snapshot tests.

The analogue of the reference's gold-file planning tests
(ref: crates/sail-spark-connect/tests/gold_data/plan/*.json, consumed by
test_gold_set): SQL -> optimized plan tree text, compared against committed
snapshots in tests/gold/plans.json. Regenerate with
`SAIL_UPDATE_GOLD=1 python -m pytest tests/test_plan_gold.py`.
"""
import json
import os

import pytest

import sail_amd
from sail_amd.engine import types as T
from sail_amd.plan import spec as S

GOLD = os.path.join(os.path.dirname(__file__), "gold", "plans.json")

CASES = {
    "filter_pushdown": "SELECT a FROM t1 JOIN t2 ON t1.k = t2.k WHERE t2.v > 5 AND a < 3",
    "join_reorder_cross": "SELECT count(*) FROM t1, t2 WHERE t1.k = t2.k AND t2.v = 1",
    "agg_split": "SELECT k, sum(a) + 1 AS s FROM t1 GROUP BY k HAVING sum(a) > 2",
    "decorrelate_exists": ("SELECT a FROM t1 WHERE EXISTS "
                           "(SELECT * FROM t2 WHERE t2.k = t1.k AND t2.v > 0)"),
    "decorrelate_scalar": ("SELECT a FROM t1 WHERE a > "
                           "(SELECT avg(v) FROM t2 WHERE t2.k = t1.k)"),
    "semi_sink": ("SELECT count(*) FROM t1 JOIN t2 ON t1.k = t2.k "
                  "WHERE t1.k IN (SELECT k FROM t3 WHERE v2 > 0)"),
    "prune_columns": "SELECT a FROM (SELECT * FROM t1 JOIN t2 ON t1.k = t2.k) x",
    "order_by_agg": "SELECT k, count(*) FROM t1 GROUP BY k ORDER BY count(*) DESC LIMIT 3",
    "pivot": "SELECT * FROM (SELECT k, v, a FROM t2 JOIN t1 USING (k)) p "
             "PIVOT (sum(a) FOR v IN (1, 2))",
    "recursive_cte": ("WITH RECURSIVE n(x) AS (SELECT 1 UNION ALL "
                      "SELECT x + 1 FROM n WHERE x < 5) SELECT sum(x) FROM n"),
    "lateral_view": ("SELECT k, e FROM t1 LATERAL VIEW explode(sequence(1, a)) "
                     "ex AS e"),
    "grouping_sets": ("SELECT k, a, count(*) FROM t1 GROUP BY ROLLUP(k, a)"),
    "window_frame": ("SELECT a, sum(a) OVER (ORDER BY a ROWS BETWEEN 1 "
                     "PRECEDING AND CURRENT ROW) FROM t1"),
}


@pytest.fixture(scope="module")
def session():
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"k": [1], "a": [1]}, schema={"k": T.I64, "a": T.I64}, name="t1")
    s.create_dataframe({"k": [1], "v": [1]}, schema={"k": T.I64, "v": T.I64}, name="t2")
    s.create_dataframe({"k": [1], "v2": [1]}, schema={"k": T.I64, "v2": T.I64}, name="t3")
    return s


def _render(s, sql):
    return S.plan_tree_string(s.plan_sql(sql)).rstrip()


def test_gold_plans(session):
    got = {name: _render(session, sql) for name, sql in CASES.items()}
    if os.environ.get("SAIL_UPDATE_GOLD") == "1" or not os.path.exists(GOLD):
        os.makedirs(os.path.dirname(GOLD), exist_ok=True)
        with open(GOLD, "w") as f:
            json.dump(got, f, indent=1, sort_keys=True)
        if os.environ.get("SAIL_UPDATE_GOLD") != "1" and os.path.exists(GOLD):
            return  # first generation
    with open(GOLD) as f:
        want = json.load(f)
    for name in CASES:
        assert got[name] == want[name], (
            f"plan drift for {name!r}:\nGOT:\n{got[name]}\nWANT:\n{want[name]}\n"
            "(if intentional: SAIL_UPDATE_GOLD=1 pytest tests/test_plan_gold.py)")
