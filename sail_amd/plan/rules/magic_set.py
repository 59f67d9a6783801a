"""Magic-set semi-filter for joined aggregates.

Pattern (q17/q20 after decorrelation):

    Join how=inner on (L.a = R.k)
      L                       -- cheap, selective (filtered part: ~0.1%)
      [Project] Aggregate keys=[k] over B   -- B huge (lineitem, 600M rows)

The aggregate computes per-key results for EVERY key in B, but the inner
join keeps only keys present in L. Semi-filter B by L's keys first:

    Aggregate.input := B SEMI JOIN (SELECT a FROM copy(L))

Correct for inner joins with any aggregate (groups eliminated by the semi
join are exactly those the join would drop). Applied only when L is a
join/aggregate-free subtree and statistically much smaller than B
(ref: the reference relies on DataFusion's decorrelation without this
rewrite; here the 600M-row whole-partition aggregate is the measured cost).
"""
from __future__ import annotations

import copy
from typing import List, Optional

from ...engine import types as T
from .. import spec as S
from .join_order import _estimate
from .util import split_conjuncts


def semi_filter_aggregates(plan: S.Plan, stats) -> S.Plan:
    if stats is None:
        return plan
    return _walk(plan, stats)


def _walk(p: S.Plan, stats) -> S.Plan:
    for attr in ("input", "left", "right"):
        c = getattr(p, attr, None)
        if isinstance(c, S.Plan):
            setattr(p, attr, _walk(c, stats))
    out = _try(p, stats)
    return out if out is not None else p


def _peel(p):
    while isinstance(p, S.SubqueryAlias):
        p = p.input
    return p


def _cheap(p: S.Plan) -> bool:
    """Join/aggregate/window-free subtree (safe to duplicate)."""
    if isinstance(p, (S.Join, S.Aggregate, S.WindowPlan, S.SetOp,
                      S.RecursiveCte, S.Generate)):
        return False
    return all(_cheap(c) for c in p.children() if c is not None)


def _try(p: S.Plan, stats) -> Optional[S.Plan]:
    if not isinstance(p, S.Join) or p.how != "inner" or p.on is None:
        return None
    for L, R, left_is_l in ((p.left, p.right, True), (p.right, p.left, False)):
        Rp = _peel(R)
        proj = None
        if isinstance(Rp, S.Project):
            proj = Rp
            Rp = _peel(Rp.input)
        if not isinstance(Rp, S.Aggregate) or not Rp.group_by or Rp.grouping_sets:
            continue
        if not _cheap(L):
            continue
        # the equi conjunct must tie an L column to an R aggregate-key output
        nl = len(p.left.schema)
        conj = split_conjuncts(p.on)
        hit = None
        for c in conj:
            if not (isinstance(c, S.BinaryOp) and c.op == "="):
                continue
            sides = [c.left, c.right]
            sides = [x.child if isinstance(x, S.Cast) else x for x in sides]
            if not all(isinstance(x, S.BoundRef) for x in sides):
                continue
            a, b = sides
            if left_is_l:
                lref = a if a.index < nl else (b if b.index < nl else None)
                rref = b if b.index >= nl else (a if a.index >= nl else None)
                if lref is None or rref is None:
                    continue
                l_local, r_local = lref.index, rref.index - nl
            else:
                lref = a if a.index >= nl else (b if b.index >= nl else None)
                rref = b if b.index < nl else (a if a.index < nl else None)
                if lref is None or rref is None:
                    continue
                l_local, r_local = lref.index - nl, rref.index
            # map r_local through the optional Project to the Aggregate output
            agg_out = r_local
            if proj is not None:
                e = proj.exprs[r_local]
                e = e.child if isinstance(e, S.Alias) else e
                if not isinstance(e, S.BoundRef):
                    continue
                agg_out = e.index
            if agg_out >= len(Rp.group_by):
                continue  # references an aggregate value, not a key
            gk = Rp.group_by[agg_out]
            gk = gk.child if isinstance(gk, S.Alias) else gk
            if not isinstance(gk, S.BoundRef):
                continue
            hit = (l_local, gk.index)
            break
        if hit is None:
            continue
        l_local, in_key = hit
        l_rows = _estimate(L, stats)[0]
        b_rows = _estimate(Rp.input, stats)[0]
        if l_rows * 20 >= b_rows:
            continue  # not selective enough to pay for the semi join
        # rewrite: Aggregate.input := input SEMI JOIN Project(copy(L), [a])
        l_copy = copy.deepcopy(L)
        keyproj = S.Project(input=l_copy, exprs=[
            S.BoundRef(l_local, L.schema[l_local][0], L.schema[l_local][1])])
        keyproj.schema = [L.schema[l_local]]
        inner = Rp.input
        ni = len(inner.schema)
        cond = S.BinaryOp("=",
                          S.BoundRef(in_key, inner.schema[in_key][0],
                                     inner.schema[in_key][1]),
                          S.BoundRef(ni, keyproj.schema[0][0],
                                     keyproj.schema[0][1]), T.BOOL)
        semi = S.Join(left=inner, right=keyproj, how="semi", on=cond)
        semi.schema = list(inner.schema)
        Rp.input = semi
        return p  # one application per join node
    return None
