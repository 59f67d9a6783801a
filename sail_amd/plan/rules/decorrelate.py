"""Subquery decorrelation.

Rewrites subquery expressions into joins — the role of the reference's
DecorrelateLateralProjection + DataFusion's decorrelation rules
(ref: crates/sail-logical-optimizer/src/lib.rs:11):

  * [NOT] EXISTS(sub)           -> left semi/anti join, correlated equality
                                   predicates lifted into the join condition
  * x [NOT] IN (sub)            -> left semi/anti join on x = sub.col
  * correlated scalar aggregate -> aggregate grouped by the correlation keys,
    (q2/q17/q20 pattern)           inner-joined back, subquery value replaced
                                   by the joined aggregate column

Equality correlation only (covers the TPC-H/ClickBench surface); anything
else raises so the failure is loud, not silently wrong.
"""
from __future__ import annotations

import copy
from typing import Dict, List, Optional, Tuple

from ...engine import types as T
from .. import spec as S
from .util import conjoin, expr_refs, outer_refs, remap_expr, split_conjuncts


class DecorrelationError(Exception):
    pass


def decorrelate(plan: S.Plan) -> S.Plan:
    """Bottom-up rewrite of every Filter whose condition holds subqueries."""
    # recurse into children first
    for attr in ("input", "left", "right"):
        child = getattr(plan, attr, None)
        if isinstance(child, S.Plan):
            setattr(plan, attr, decorrelate(child))
    if isinstance(plan, S.WithCte):
        plan.ctes = [(n, decorrelate(p)) for n, p in plan.ctes]
    # subqueries inside expressions (uncorrelated scalar subqueries stay;
    # their *plans* still need decorrelation inside)
    for e in _plan_exprs(plan):
        _decorrelate_nested(e)

    if isinstance(plan, S.Filter):
        key = plan.__dict__.get("_cte_cache_key")
        out = _rewrite_filter(plan)
        if key is not None:
            out.__dict__["_cte_cache_key"] = key
        return out
    return plan


def _plan_exprs(p: S.Plan):
    if isinstance(p, S.Project):
        return p.exprs
    if isinstance(p, S.Filter):
        return [p.condition]
    if isinstance(p, S.Join) and p.on is not None:
        return [p.on]
    if isinstance(p, S.Aggregate):
        return list(p.group_by) + list(p.aggs)
    return []


def _decorrelate_nested(e: S.Expr):
    if isinstance(e, (S.ScalarSubquery, S.Exists)):
        e.plan = decorrelate(e.plan)
        return
    if isinstance(e, S.InSubquery):
        e.plan = decorrelate(e.plan)
        _decorrelate_nested(e.child)
        return
    for c in e.children():
        _decorrelate_nested(c)


def _contains_subquery(e: S.Expr) -> bool:
    if isinstance(e, (S.ScalarSubquery, S.Exists, S.InSubquery)):
        return True
    return any(_contains_subquery(c) for c in e.children())


def _contains_correlated_scalar(e: S.Expr) -> bool:
    if isinstance(e, S.ScalarSubquery):
        return bool(outer_refs(e.plan))
    return any(_contains_correlated_scalar(c) for c in e.children())


def _rewrite_filter(f: S.Filter) -> S.Plan:
    base = f.input
    orig_n = len(base.schema)
    conds = split_conjuncts(f.condition)
    remaining: List[S.Expr] = []
    current: S.Plan = base

    for c in conds:
        if isinstance(c, S.Exists):
            current = _apply_exists(current, c.plan, negated=c.negated)
        elif isinstance(c, S.UnaryOp) and c.op == "not" and isinstance(c.child, S.Exists):
            current = _apply_exists(current, c.child.plan, negated=not c.child.negated)
        elif isinstance(c, S.InSubquery):
            current = _apply_in(current, c)
        elif isinstance(c, S.UnaryOp) and c.op == "not" and isinstance(c.child, S.InSubquery):
            inner = c.child
            current = _apply_in(current, S.InSubquery(inner.child, inner.plan,
                                                      not inner.negated, T.BOOL))
        elif _contains_correlated_scalar(c):
            current, c2 = _apply_correlated_scalar(current, c)
            remaining.append(c2)
        else:
            remaining.append(c)

    out: S.Plan = current
    cond = conjoin(remaining)
    if cond is not None:
        flt = S.Filter(input=out, condition=cond)
        flt.schema = out.schema
        out = flt
    if len(out.schema) != orig_n:
        from .util import make_project

        out = make_project(out, list(range(orig_n)))
    return out


# ---------------------------------------------------------------------------

def _lift_correlation(sub: S.Plan) -> Tuple[S.Plan, List[Tuple[int, S.Expr]], List[S.Expr]]:
    """Remove correlated predicates from Filters inside `sub`.

    Returns (new_sub, equi, residual) where:
      equi:     [(outer_index, sub_expr)] equality pairs outer_col = sub_expr
                — sub_expr is bound against new_sub's *output* schema
      residual: other correlated predicates rewritten with OuterRef kept
                (resolved later against the join pair scope)

    Only handles correlation inside Filter nodes whose path to the sub root
    consists of column-preserving operators (Filter/SubqueryAlias) or a
    trailing Project/Aggregate handled by the callers.
    """
    equi: List[Tuple[int, S.Expr]] = []
    residual: List[S.Expr] = []

    def is_corr(e: S.Expr) -> bool:
        return any(isinstance(x, S.OuterRef) for x in _walk(e))

    def strip(p: S.Plan) -> S.Plan:
        if isinstance(p, S.Filter):
            inner = strip(p.input)
            keep = []
            for c in split_conjuncts(p.condition):
                if not is_corr(c):
                    keep.append(c)
                    continue
                pair = _as_outer_equality(c)
                if pair is not None:
                    equi.append(pair)
                else:
                    residual.append(c)
            cond = conjoin(keep)
            if cond is None:
                return inner
            out = S.Filter(input=inner, condition=cond)
            out.schema = inner.schema
            return out
        if isinstance(p, S.SubqueryAlias):
            inner = strip(p.input)
            out = S.SubqueryAlias(input=inner, alias=p.alias, column_aliases=p.column_aliases)
            out.schema = p.schema
            return out
        if isinstance(p, S.Join):
            # correlation may live inside a join input (q2's nested join tree)
            left = strip(p.left)
            right0 = len(p.left.schema)
            # note: stripping below a join keeps indices valid because strip
            # never changes schemas
            right = strip(p.right)
            out = S.Join(left=left, right=right, how=p.how, on=p.on, using=p.using)
            out.schema = p.schema
            return out
        if isinstance(p, S.Project):
            inner = strip(p.input)
            out = S.Project(input=inner, exprs=p.exprs)
            out.schema = p.schema
            return out
        return p

    new_sub = strip(sub)
    return new_sub, equi, residual


def _walk(e: S.Expr):
    yield e
    for c in e.children():
        yield from _walk(c)


def _as_outer_equality(c: S.Expr) -> Optional[Tuple[int, S.Expr]]:
    """Match OuterRef = local_expr (either side)."""
    if isinstance(c, S.BinaryOp) and c.op == "=":
        l, r = c.left, c.right
        l = l.child if isinstance(l, S.Cast) else l
        r = r.child if isinstance(r, S.Cast) else r
        if isinstance(l, S.OuterRef) and not any(isinstance(x, S.OuterRef) for x in _walk(r)):
            return (l.index, r)
        if isinstance(r, S.OuterRef) and not any(isinstance(x, S.OuterRef) for x in _walk(l)):
            return (r.index, l)
    return None


def _sub_expr_to_output(sub: S.Plan, e: S.Expr) -> Optional[int]:
    """Map an expression bound against the *internals* of `sub` to an output
    ordinal of `sub`, appending a passthrough column when the sub's top is a
    Project that doesn't already expose it."""
    if isinstance(e, S.BoundRef):
        return e.index
    return None


def _apply_exists(current: S.Plan, sub: S.Plan, negated: bool) -> S.Plan:
    """current [anti|semi] JOIN sub' ON lifted-correlated-predicates."""
    sub2, equi, residual = _lift_correlation(copy.deepcopy(sub))
    sub2 = _expose_for_join(sub2, equi, residual)
    nleft = len(current.schema)
    conds: List[S.Expr] = []
    for k, (outer_idx, local) in enumerate(_equi_pairs(sub2)):
        lref = S.BoundRef(outer_idx, current.schema[outer_idx][0], current.schema[outer_idx][1])
        rref = S.BoundRef(nleft + k, f"__corr{k}", local.dtype)
        conds.append(S.BinaryOp("=", lref, rref, T.BOOL))
    for rc in _residual_conds(sub2):
        conds.append(_rebind_residual(rc, nleft, current, sub2))
    how = "anti" if negated else "semi"
    out = S.Join(left=current, right=sub2, how=how, on=conjoin(conds), using=None)
    out.schema = list(current.schema)
    return out


def _apply_in(current: S.Plan, e: S.InSubquery) -> S.Plan:
    sub2, equi, residual = _lift_correlation(copy.deepcopy(e.plan))
    if len(sub2.schema) != 1:
        raise DecorrelationError("IN subquery must produce one column")
    sub2 = _expose_for_join(sub2, equi, residual, keep_first=True)
    nleft = len(current.schema)
    conds: List[S.Expr] = [S.BinaryOp("=", e.child,
                                      S.BoundRef(nleft + 0, sub2.schema[0][0], sub2.schema[0][1]),
                                      T.BOOL)]
    k0 = 1
    for k, (outer_idx, local) in enumerate(_equi_pairs(sub2)):
        lref = S.BoundRef(outer_idx, current.schema[outer_idx][0], current.schema[outer_idx][1])
        rref = S.BoundRef(nleft + k0 + k, f"__corr{k}", local.dtype)
        conds.append(S.BinaryOp("=", lref, rref, T.BOOL))
    for rc in _residual_conds(sub2):
        conds.append(_rebind_residual(rc, nleft, current, sub2))
    how = "anti" if e.negated else "semi"
    out = S.Join(left=current, right=sub2, how=how, on=conjoin(conds), using=None)
    out.schema = list(current.schema)
    return out


def _apply_correlated_scalar(current: S.Plan, cond: S.Expr) -> Tuple[S.Plan, S.Expr]:
    """Rewrite a predicate containing a correlated scalar aggregate subquery:
    join `current` with the grouped aggregate and substitute the value.
    Pattern: sub = Project[expr_over_agg](Aggregate(no groups, Filter(corr)))."""
    # find the subquery node
    holder: List[S.ScalarSubquery] = []

    def find(e: S.Expr):
        if isinstance(e, S.ScalarSubquery) and outer_refs(e.plan):
            holder.append(e)
            return
        for c in e.children():
            find(c)

    find(cond)
    if not holder:
        return current, cond
    sq = holder[0]
    sub = copy.deepcopy(sq.plan)
    # expect Project over Aggregate
    if not (isinstance(sub, S.Project) and isinstance(sub.input, S.Aggregate)
            and not sub.input.group_by):
        raise DecorrelationError(
            f"unsupported correlated scalar subquery shape: {type(sub).__name__}")
    agg: S.Aggregate = sub.input
    inner, equi, residual = _lift_correlation(agg.input)
    if residual:
        raise DecorrelationError("non-equality correlation in scalar subquery")
    if not equi:
        raise DecorrelationError("scalar subquery marked correlated but no equality found")
    # build grouped aggregate: group by correlation keys
    group_exprs = [local for _, local in equi]
    agg2 = S.Aggregate(input=inner, group_by=group_exprs, aggs=agg.aggs)
    agg2.schema = ([(f"__ck{i}", g.dtype) for i, g in enumerate(group_exprs)]
                   + [(f"__agg{i}", a.dtype) for i, a in enumerate(agg.aggs)])
    # project: correlation keys + the sub's output expression (rebased)
    shift = len(group_exprs) - 0
    # original project exprs reference agg schema: [agg0, agg1...] at positions
    # 0..n-1 -> in agg2 they live at positions len(groups)..; remap
    remap = {i: i + len(group_exprs) for i in range(len(agg.aggs))}
    val_expr = remap_expr(sub.exprs[0], remap)
    proj_exprs = [S.BoundRef(i, f"__ck{i}", g.dtype) for i, g in enumerate(group_exprs)] \
        + [val_expr]
    proj = S.Project(input=agg2, exprs=proj_exprs)
    proj.schema = [(f"__ck{i}", g.dtype) for i, g in enumerate(group_exprs)] \
        + [("__sqval", val_expr.dtype)]
    # join current with proj on outer keys
    nleft = len(current.schema)
    conds = []
    for k, (outer_idx, _local) in enumerate(equi):
        lref = S.BoundRef(outer_idx, current.schema[outer_idx][0], current.schema[outer_idx][1])
        rref = S.BoundRef(nleft + k, f"__ck{k}", proj.schema[k][1])
        conds.append(S.BinaryOp("=", lref, rref, T.BOOL))
    join = S.Join(left=current, right=proj, how="inner", on=conjoin(conds), using=None)
    join.schema = list(current.schema) + list(proj.schema)
    # replace subquery with ref to __sqval
    val_ref = S.BoundRef(nleft + len(equi), "__sqval", val_expr.dtype)

    def replace(e: S.Expr) -> S.Expr:
        if e is sq or (isinstance(e, S.ScalarSubquery) and e.plan is sq.plan):
            return val_ref
        ch = e.children()
        if not ch:
            return e
        out = e.with_children([replace(c) for c in ch])
        out.dtype = e.dtype
        return out

    return join, replace(cond)


# -- helpers for exists/in join construction --------------------------------

def _expose_for_join(sub: S.Plan, equi, residual, keep_first: bool = False) -> S.Plan:
    """Wrap `sub` in a projection exposing [first col?] + correlation local
    exprs + residual-referenced local columns; stashes metadata on the node."""
    exprs: List[S.Expr] = []
    schema: List[Tuple[str, T.DataType]] = []
    if keep_first:
        exprs.append(S.BoundRef(0, sub.schema[0][0], sub.schema[0][1]))
        schema.append(sub.schema[0])
    for k, (outer_idx, local) in enumerate(equi):
        exprs.append(local)
        schema.append((f"__corr{k}", local.dtype))
    # residual predicates may reference arbitrary local columns: expose all
    # columns after the correlation keys (simple and correct; pruning trims)
    base_cols = len(exprs)
    for i, (n, t) in enumerate(sub.schema):
        exprs.append(S.BoundRef(i, n, t))
        schema.append((n, t))
    proj = S.Project(input=sub, exprs=exprs)
    proj.schema = schema
    proj.__dict__["_equi"] = [(outer_idx, local) for outer_idx, local in equi]
    proj.__dict__["_residual"] = residual
    proj.__dict__["_keep_first"] = keep_first
    proj.__dict__["_base_cols"] = base_cols
    return proj


def _equi_pairs(sub2: S.Plan):
    return sub2.__dict__.get("_equi", [])


def _residual_conds(sub2: S.Plan):
    return sub2.__dict__.get("_residual", [])


def _rebind_residual(rc: S.Expr, nleft: int, current: S.Plan, sub2: S.Plan) -> S.Expr:
    """Residual correlated predicate: OuterRef i -> left side index i;
    local BoundRef j -> right side at nleft + base_cols + j."""
    base = sub2.__dict__.get("_base_cols", 0)

    def rb(e: S.Expr) -> S.Expr:
        if isinstance(e, S.OuterRef):
            return S.BoundRef(e.index, e.name, e.dtype)
        if isinstance(e, S.BoundRef):
            return S.BoundRef(nleft + base + e.index, e.name, e.dtype)
        ch = e.children()
        if not ch:
            return e
        out = e.with_children([rb(c) for c in ch])
        out.dtype = e.dtype
        return out

    return rb(rc)
