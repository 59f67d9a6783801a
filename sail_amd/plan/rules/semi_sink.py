"""Sink semi/anti joins below inner joins.

Decorrelation leaves semi/anti joins ABOVE the whole FROM-clause join tree
(the filter they replaced sat there). When the join condition references only
one input of an inner join below, the semi/anti join can apply directly to
that input — turning "join everything, then filter orders by the subquery"
into "filter orders first" (q4/q18/q21/q22 shape).

Semi/anti output schema equals its left input's schema, so sinking onto the
left inner input keeps indices; onto the right input they shift by -nleft.
Runs after join reordering, before pruning.
"""
from __future__ import annotations

from ...engine import types as T
from .. import spec as S
from .util import expr_refs, remap_expr


def sink_semi_joins(plan: S.Plan) -> S.Plan:
    for attr in ("input", "left", "right"):
        child = getattr(plan, attr, None)
        if isinstance(child, S.Plan):
            setattr(plan, attr, sink_semi_joins(child))
    for e in _exprs(plan):
        _walk_expr(e)
    if isinstance(plan, S.Join) and plan.how in ("semi", "anti"):
        return _sink(plan)
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


def _walk_expr(e: S.Expr):
    if isinstance(e, (S.ScalarSubquery, S.Exists)):
        e.plan = sink_semi_joins(e.plan)
        return
    if isinstance(e, S.InSubquery):
        e.plan = sink_semi_joins(e.plan)
    for c in e.children():
        _walk_expr(c)


def _sink(semi: S.Join) -> S.Plan:
    target = semi.left
    if semi.__dict__.get("_cte_cache_key") is not None:
        return semi
    if not isinstance(target, S.Join) or target.how not in ("inner", "cross"):
        return semi
    nleft = len(target.left.schema)
    ntarget = len(target.schema)
    # indices < ntarget reference the target (left side of the semi);
    # >= ntarget reference the sub (right of semi) and do not block sinking
    left_refs = {i for i in expr_refs(semi.on) if i < ntarget} if semi.on is not None else set()
    if left_refs and all(i < nleft for i in left_refs):
        inner_left = target.left
        # shift sub-side references by the width change (target -> inner_left)
        delta = len(inner_left.schema) - ntarget
        on = remap_expr(semi.on, {i: (i if i < nleft else i + delta)
                                  for i in expr_refs(semi.on)})
        new_semi = S.Join(left=inner_left, right=semi.right, how=semi.how,
                          on=on, using=None)
        new_semi.schema = inner_left.schema
        sunk = _sink(new_semi)
        out = S.Join(left=sunk, right=target.right, how=target.how,
                     on=target.on, using=target.using)
        out.schema = target.schema
        return out
    if left_refs and all(i >= nleft for i in left_refs):
        inner_right = target.right
        delta = len(inner_right.schema) - ntarget
        on = remap_expr(semi.on, {i: (i - nleft if i < ntarget else i + delta)
                                  for i in expr_refs(semi.on)})
        new_semi = S.Join(left=inner_right, right=semi.right, how=semi.how,
                          on=on, using=None)
        new_semi.schema = inner_right.schema
        sunk = _sink(new_semi)
        out = S.Join(left=target.left, right=sunk, how=target.how,
                     on=target.on, using=target.using)
        out.schema = target.schema
        return out
    return semi
