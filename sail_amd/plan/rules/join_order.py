"""Join reordering.

Flattens maximal inner/cross-join trees into (leaves, conjuncts), estimates
leaf cardinalities with selectivity heuristics, and rebuilds a left-deep tree
greedily: start from the smallest filtered leaf, repeatedly join the
edge-connected leaf minimizing the estimated intermediate size. The greedy
counterpart of the reference's DP join reorderer
(ref: crates/sail-physical-optimizer/src/join_reorder/mod.rs:31 — graph
builder + cardinality estimator + cost model; DP upgrade is planned).

Build-side note: the executor builds the hash table on the RIGHT input, so
the accumulated (large) side stays on the left as probe.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Set, Tuple

from ...engine import types as T
from .. import spec as S
from .util import conjoin, expr_refs, remap_expr, split_conjuncts


def reorder_joins(plan: S.Plan, stats: Optional[Callable[[str], Optional[int]]] = None) -> S.Plan:
    return _walk(plan, stats)


def _walk(plan: S.Plan, stats) -> S.Plan:
    # rewrite subquery plans too
    for e in _exprs(plan):
        _walk_expr(e, stats)
    key = plan.__dict__.get("_cte_cache_key")
    if isinstance(plan, S.Join) and plan.how in ("inner", "cross"):
        out = _reorder_tree(plan, stats)
        if key is not None:
            out.__dict__["_cte_cache_key"] = key
        return out
    if False:
        return _reorder_tree(plan, stats)
    for attr in ("input", "left", "right"):
        child = getattr(plan, attr, None)
        if isinstance(child, S.Plan):
            setattr(plan, attr, _walk(child, stats))
    return plan


def _exprs(p: S.Plan):
    if isinstance(p, S.Project):
        return p.exprs
    if isinstance(p, S.Filter):
        return [p.condition]
    if isinstance(p, S.Join) and p.on is not None:
        return [p.on]
    if isinstance(p, S.Aggregate):
        return list(p.group_by) + list(p.aggs)
    return []


def _walk_expr(e: S.Expr, stats):
    if isinstance(e, (S.ScalarSubquery, S.Exists)):
        e.plan = _walk(e.plan, stats)
        return
    if isinstance(e, S.InSubquery):
        e.plan = _walk(e.plan, stats)
    for c in e.children():
        _walk_expr(c, stats)


# ---------------------------------------------------------------------------

def _flatten(plan: S.Plan, leaves: List[S.Plan], conds: List[S.Expr], offset: int) -> int:
    """Collect inner-join leaves in order; returns total width. Conjunct
    indices are valid in the concatenation of leaf schemas (schema concat is
    associative for inner joins)."""
    if isinstance(plan, S.Join) and plan.how in ("inner", "cross") \
            and plan.__dict__.get("_cte_cache_key") is None:
        lw = _flatten(plan.left, leaves, conds, offset)
        rw = _flatten(plan.right, leaves, conds, offset + lw)
        if plan.on is not None:
            for c in split_conjuncts(plan.on):
                conds.append(remap_expr(c, {i: i + offset for i in expr_refs(c)}))
        return lw + rw
    leaves.append(plan)
    return len(plan.schema)


def _estimate(plan: S.Plan, stats) -> Tuple[float, float]:
    """(estimated rows, base rows) for a leaf subtree."""
    if isinstance(plan, S.Read):
        n = None
        if stats is not None:
            n = stats.table_rows(plan.table) if hasattr(stats, "table_rows") else stats(plan.table)
        base = float(n) if n else 1000.0
        return base, base
    if isinstance(plan, S.DataSourceRead):
        # scan views stamp the catalog name on the node (_table_name)
        name = plan.__dict__.get("_table_name")
        n = stats.table_rows(name) if (name and stats is not None
                                       and hasattr(stats, "table_rows")) else None
        base = float(n) if n else 1000.0
        return base, base
    if isinstance(plan, S.SubqueryAlias):
        # scan views (register_tpch_parquet / CREATE VIEW over a path):
        # the alias IS the catalog name carrying row stats — without this
        # the reorderer is blind behind DataSourceRead and q5-style plans
        # explode (measured: 893 GiB probe at SF100)
        inner = plan.input
        while isinstance(inner, S.Project):
            inner = inner.input
        if isinstance(inner, S.DataSourceRead) and stats is not None \
                and hasattr(stats, "table_rows"):
            n = stats.table_rows(plan.alias)
            if n:
                return float(n), float(n)
    if isinstance(plan, (S.SubqueryAlias, S.Project, S.Limit, S.Sort)):
        child = plan.input
        est, base = _estimate(child, stats)
        if isinstance(plan, S.Limit) and plan.n is not None:
            est = min(est, float(plan.n))
        return est, base
    if isinstance(plan, S.Filter):
        est, base = _estimate(plan.input, stats)
        sel = 1.0
        for c in split_conjuncts(plan.condition):
            sel *= _selectivity(c)
        return max(est * sel, 1.0), base
    if isinstance(plan, S.Aggregate):
        est, base = _estimate(plan.input, stats)
        if not plan.group_by:
            return 1.0, 1.0
        return max(est / 10.0, 1.0), base
    if isinstance(plan, S.Join):
        le, lb = _estimate(plan.left, stats)
        re_, rb = _estimate(plan.right, stats)
        if plan.how in ("semi", "anti"):
            return max(le * 0.5, 1.0), lb
        denom = max(lb, rb, 1.0)
        return max(le * re_ / denom, 1.0), max(lb, rb)
    if isinstance(plan, (S.LocalRelation, S.Range)):
        n = len(next(iter(plan.data.values()))) if isinstance(plan, S.LocalRelation) and plan.data else 10
        if isinstance(plan, S.Range):
            n = max((plan.end - plan.start) // max(plan.step, 1), 1)
        return float(n), float(n)
    if isinstance(plan, S.SetOp):
        le, lb = _estimate(plan.left, stats)
        re_, rb = _estimate(plan.right, stats)
        return le + re_, lb + rb
    if isinstance(plan, S.Distinct):
        est, base = _estimate(plan.input, stats)
        return max(est / 2.0, 1.0), base
    return 1000.0, 1000.0


def _selectivity(c: S.Expr) -> float:
    if isinstance(c, S.BinaryOp):
        if c.op == "=":
            return 0.05
        if c.op in ("<", "<=", ">", ">="):
            return 0.33
        if c.op == "!=":
            return 0.9
        if c.op == "or":
            return min(0.9, _selectivity(c.left) + _selectivity(c.right))
        if c.op == "and":
            return _selectivity(c.left) * _selectivity(c.right)
    if isinstance(c, S.Between):
        return 0.15
    if isinstance(c, S.InList):
        return min(0.9, 0.05 * max(len(c.values), 1))
    if isinstance(c, S.Like):
        return 0.1 if not c.negated else 0.9
    if isinstance(c, S.UnaryOp) and c.op in ("isnull",):
        return 0.05
    return 0.5


def _leaf_column_source(leaf: S.Plan, k: int):
    """Trace output column k of a leaf subtree to (table_name, column_name)
    if it is a direct base-table column (including scan views: the enclosing
    SubqueryAlias name keys the catalog statistics)."""
    p = leaf
    alias = None
    while True:
        if isinstance(p, S.SubqueryAlias):
            alias = p.alias or alias
            p = p.input
            continue
        if isinstance(p, (S.Filter, S.Limit, S.Sort, S.Distinct)):
            p = p.input
            continue
        if isinstance(p, S.Project):
            e = p.exprs[k]
            if isinstance(e, S.Alias):
                e = e.child
            if isinstance(e, S.BoundRef):
                k = e.index
                p = p.input
                continue
            return None
        if isinstance(p, S.Read):
            if 0 <= k < len(p.schema):
                return (p.table, p.schema[k][0])
            return None
        if isinstance(p, S.DataSourceRead):
            name = p.__dict__.get("_table_name") or alias
            if name and 0 <= k < len(p.schema or []):
                return (name, p.schema[k][0])
            return None
        return None


def _conjunct_ndv(c: S.Expr, leaves, offs, leaf_of, ests, stats) -> float:
    """ndv denominator for one equality conjunct between two leaves."""
    best = None
    if isinstance(c, S.BinaryOp) and c.op == "=":
        for side in (c.left, c.right):
            e = side.child if isinstance(side, S.Cast) else side
            if not isinstance(e, S.BoundRef):
                continue
            li = leaf_of(e.index)
            col_k = e.index - offs[li]
            ndv = None
            if stats is not None and hasattr(stats, "column_stats"):
                srcinfo = _leaf_column_source(leaves[li], col_k)
                if srcinfo is not None:
                    cs = stats.column_stats(srcinfo[0], srcinfo[1])
                    if cs is not None and cs[1] is not None:
                        ndv = float(cs[1])
            if ndv is None:
                ndv = ests[li][1]  # base rows fallback
            best = ndv if best is None else max(best, ndv)
    return best if best is not None else 1000.0


def _reorder_tree(root: S.Join, stats) -> S.Plan:
    leaves: List[S.Plan] = []
    conds: List[S.Expr] = []
    _flatten(root, leaves, conds, 0)
    n = len(leaves)
    if n > 12:
        # beyond reorder budget: keep original shape but recurse into leaves
        for i, lf in enumerate(leaves):
            leaves[i] = _walk(lf, stats)
        return root
    leaves = [_walk(lf, stats) for lf in leaves]

    # old concatenated offsets
    offs = [0]
    for lf in leaves:
        offs.append(offs[-1] + len(lf.schema))

    def leaf_of(idx: int) -> int:
        for li in range(n):
            if offs[li] <= idx < offs[li + 1]:
                return li
        raise IndexError(idx)

    ests = [_estimate(lf, stats) for lf in leaves]

    # edge conjuncts with per-key ndv estimates from base-table stats
    edges: Dict[int, Set[int]] = {i: set() for i in range(n)}
    edge_ndv: List[Tuple[int, int, float]] = []  # (leaf_a, leaf_b, denom)
    for c in conds:
        ls = {leaf_of(i) for i in expr_refs(c)}
        if len(ls) == 2:
            a, b = sorted(ls)
            edges[a].add(b)
            edges[b].add(a)
            denom = _conjunct_ndv(c, leaves, offs, leaf_of, ests, stats)
            edge_ndv.append((a, b, denom))

    def join_denom(j: int, placed: Set[int], cur_est: float) -> float:
        """Combined ndv for all equality conjuncts connecting leaf j to the
        placed set. Per-key ndvs multiply but the composite-key ndv can never
        exceed either side's cardinality (a table's PK tuple has ndv = its
        row count, e.g. partsupp's (partkey, suppkey))."""
        d = 1.0
        for a, b, nd in edge_ndv:
            if (a == j and b in placed) or (b == j and a in placed):
                d = min(d * nd, 1e15)
        return min(d, max(ests[j][1], cur_est, 1.0))

    def step_cost(i: int, placed: Set[int], cur_est: float, cur_sel: float) -> float:
        """Estimated rows after joining leaf i onto the placed set — shared
        by the greedy and the DP search so both optimize one model."""
        est_i, base_i = ests[i]
        if any(j in placed for j in edges[i]):
            formula = cur_est * est_i / max(join_denom(i, placed, cur_est), 1.0)
            return max(formula, est_i * cur_sel)
        return cur_est * est_i * (1.0 if est_i <= 2 else 8.0)

    def run_dp():
        """Exact DP over left-deep orders (all subsets, best-total state per
        subset) — the DP counterpart of the reference's join_reorder DP
        (ref: crates/sail-physical-optimizer/src/join_reorder/mod.rs:31).
        O(2^n * n^2); used for n <= 10, multi-seed greedy beyond."""
        dp: Dict[int, Tuple[float, float, float, Tuple[int, ...]]] = {}
        for i in range(n):
            sel = min(1.0, ests[i][0] / max(ests[i][1], 1.0))
            dp[1 << i] = (ests[i][0], ests[i][0], sel, (i,))
        full = (1 << n) - 1
        masks = sorted(dp.keys())
        by_pop: Dict[int, List[int]] = {1: masks}
        for pop in range(1, n):
            nxt: Dict[int, Tuple[float, float, float, Tuple[int, ...]]] = {}
            for mask in by_pop[pop]:
                total, cur_est, cur_sel, order = dp[mask]
                placed = {order[k] for k in range(len(order))}
                for i in range(n):
                    bit = 1 << i
                    if mask & bit:
                        continue
                    c = max(step_cost(i, placed, cur_est, cur_sel), 1.0)
                    est_b, base_b = ests[i]
                    nsel = min(cur_sel, min(1.0, est_b / max(base_b, 1.0)))
                    nt = total + c
                    m2 = mask | bit
                    old = nxt.get(m2) or dp.get(m2)
                    if old is None or nt < old[0]:
                        nxt[m2] = (nt, c, nsel, order + (i,))
            dp.update(nxt)
            by_pop[pop + 1] = [m for m in nxt]
        if full not in dp:
            return None, None
        total, _, _, order = dp[full]
        return list(order), total

    # greedy from EVERY seed, objective = sum of intermediate sizes; a single
    # smallest-leaf seed can dead-end into a low-ndv edge (q5: region ->
    # nation -> supplier forces customer x supplier on nationkey)
    def run_greedy(seed: int):
        order = [seed]
        placed = {seed}
        cur_est = ests[seed][0]
        # selectivity of the placed side: FK joins against an unfiltered fact
        # keep every fact row scaled by this (containment lower bound) —
        # without it, partsupp⋈lineitem estimates 1e5 instead of 6e8 (q9)
        cur_sel = min(1.0, ests[seed][0] / max(ests[seed][1], 1.0))
        total = cur_est
        while len(placed) < n:
            best, best_cost = None, None
            for i in range(n):
                if i in placed:
                    continue
                est_i, base_i = ests[i]
                if any(j in placed for j in edges[i]):
                    formula = cur_est * est_i / max(join_denom(i, placed, cur_est), 1.0)
                    cost = max(formula, est_i * cur_sel)
                else:
                    # disconnected cross product: greedy is myopic, so a
                    # cheap-looking cross (1-row dims aside) poisons later
                    # steps — penalize unless genuinely tiny
                    cost = cur_est * est_i * (1.0 if est_i <= 2 else 8.0)
                if best_cost is None or cost < best_cost:
                    best, best_cost = i, cost
            order.append(best)
            placed.add(best)
            est_b, base_b = ests[best]
            cur_sel = min(cur_sel, min(1.0, est_b / max(base_b, 1.0)))
            cur_est = max(best_cost, 1.0)
            total += cur_est
        return order, total

    best_order, best_total = None, None
    for seed in range(n):
        order_s, total_s = run_greedy(seed)
        if best_total is None or total_s < best_total:
            best_order, best_total = order_s, total_s
    if n <= 10:
        order_dp, total_dp = run_dp()
        if order_dp is not None and total_dp < best_total:
            best_order, best_total = order_dp, total_dp
    order = best_order

    # rebuild left-deep tree in `order`, remapping conjunct indices
    new_off: Dict[int, int] = {}
    pos = 0
    for li in order:
        new_off[li] = pos
        pos += len(leaves[li].schema)

    def remap_cond(c: S.Expr) -> S.Expr:
        mapping = {}
        for i in expr_refs(c):
            li = leaf_of(i)
            mapping[i] = new_off[li] + (i - offs[li])
        return remap_expr(c, mapping)

    remaining = [(c, {leaf_of(i) for i in expr_refs(c)}, remap_cond(c)) for c in conds]
    cur = leaves[order[0]]
    avail = {order[0]}
    avail_width = len(cur.schema)
    for li in order[1:]:
        right = leaves[li]
        avail.add(li)
        place_now = [rc for c, ls, rc in remaining if ls and ls.issubset(avail)]
        remaining = [(c, ls, rc) for c, ls, rc in remaining if not (ls and ls.issubset(avail))]
        on = conjoin(place_now)
        j = S.Join(left=cur, right=right, how="inner" if on is not None else "cross",
                   on=on, using=None)
        j.schema = list(cur.schema) + list(right.schema)
        cur = j
        avail_width += len(right.schema)
    # degenerate conjuncts (no refs / single-leaf leftovers) become a filter
    leftover = [rc for c, ls, rc in remaining]
    if leftover:
        f = S.Filter(input=cur, condition=conjoin(leftover))
        f.schema = cur.schema
        cur = f

    # restore original column order with a projection
    perm = []
    for li in range(n):
        for k in range(len(leaves[li].schema)):
            perm.append(new_off[li] + k)
    if perm != list(range(len(perm))):
        pr = S.Project(input=cur, exprs=[
            S.BoundRef(p, cur.schema[p][0], cur.schema[p][1]) for p in perm])
        pr.schema = [cur.schema[p] for p in perm]
        cur = pr
    return cur
