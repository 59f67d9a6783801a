"""Shared rewrite utilities for optimizer rules (bound-tree index algebra)."""
from __future__ import annotations

import copy
from typing import Dict, List, Optional, Set

from ...engine import types as T
from .. import spec as S


def expr_refs(e: S.Expr) -> Set[int]:
    """Set of input ordinals referenced by a bound expression (excluding
    subquery-internal references)."""
    out: Set[int] = set()
    _collect_refs(e, out)
    return out


def _collect_refs(e: S.Expr, out: Set[int]):
    if e is None:
        return
    if isinstance(e, S.BoundRef):
        out.add(e.index)
        return
    if isinstance(e, (S.ScalarSubquery, S.Exists)):
        return  # internal plan has its own scope
    if isinstance(e, S.InSubquery):
        _collect_refs(e.child, out)
        return
    if isinstance(e, S.Lambda):
        # body refs 0..k-1 are lambda params; k+i is enclosing-chunk col i
        k = len(e.params)
        inner: Set[int] = set()
        _collect_refs(e.body, inner)
        out.update(r - k for r in inner if r >= k)
        return
    for c in e.children():
        _collect_refs(c, out)


def outer_refs(plan: S.Plan) -> List[S.OuterRef]:
    """All OuterRef nodes appearing anywhere in a plan subtree."""
    out: List[S.OuterRef] = []

    def walk_expr(e):
        if e is None:
            return
        if isinstance(e, S.OuterRef):
            out.append(e)
            return
        if isinstance(e, (S.ScalarSubquery, S.Exists)):
            walk_plan(e.plan)
            return
        if isinstance(e, S.InSubquery):
            walk_expr(e.child)
            walk_plan(e.plan)
            return
        for c in e.children():
            walk_expr(c)

    def walk_plan(p):
        for e in plan_exprs(p):
            walk_expr(e)
        for c in p.children():
            walk_plan(c)

    walk_plan(plan)
    return out


def plan_exprs(p: S.Plan) -> List[S.Expr]:
    if isinstance(p, S.Project):
        return list(p.exprs)
    if isinstance(p, S.Filter):
        return [p.condition]
    if isinstance(p, S.Join):
        return [p.on] if p.on is not None else []
    if isinstance(p, S.Aggregate):
        return list(p.group_by) + list(p.aggs)
    if isinstance(p, S.Sort):
        return list(p.keys)
    if isinstance(p, S.WindowPlan):
        return list(p.window_exprs)
    return []


def remap_expr(e: S.Expr, mapping: Dict[int, int]) -> S.Expr:
    """Rewrite BoundRef ordinals through `mapping` (copying the tree)."""
    if e is None:
        return None
    if isinstance(e, S.BoundRef):
        return S.BoundRef(mapping[e.index], e.name, e.dtype)
    if isinstance(e, S.ScalarSubquery):
        return e
    if isinstance(e, S.Exists):
        return e
    if isinstance(e, S.InSubquery):
        return S.InSubquery(remap_expr(e.child, mapping), e.plan, e.negated, e.dtype)
    if isinstance(e, S.Alias):
        return S.Alias(remap_expr(e.child, mapping), e.name, e.dtype)
    if isinstance(e, S.AggFunc):
        return S.AggFunc(e.name, [remap_expr(a, mapping) for a in e.args], e.distinct, e.dtype,
                         remap_expr(e.filter, mapping) if e.filter is not None else None)
    if isinstance(e, S.SortKey):
        return S.SortKey(remap_expr(e.child, mapping), e.ascending, e.nulls_first)
    if isinstance(e, S.Lambda):
        k = len(e.params)
        inner_map = {j: j for j in range(k)}
        inner_map.update({i + k: v + k for i, v in mapping.items()})
        return S.Lambda(e.params, remap_expr(e.body, inner_map), e.dtype)
    if isinstance(e, S.WindowExpr):
        return S.WindowExpr(func=remap_expr(e.func, mapping),
                            partition_by=[remap_expr(x, mapping) for x in e.partition_by],
                            order_by=[remap_expr(k, mapping) for k in e.order_by],
                            frame=e.frame, dtype=e.dtype)
    ch = e.children()
    if not ch:
        return e
    out = e.with_children([remap_expr(c, mapping) for c in ch])
    out.dtype = e.dtype
    return out


def substitute_refs(e: S.Expr, exprs: List[S.Expr]) -> S.Expr:
    """Replace BoundRef i with exprs[i] (push an expr through a projection)."""
    if e is None:
        return None
    if isinstance(e, S.Lambda):
        # body refs 0..k-1 are lambda params, k+i is enclosing col i: keep
        # params, substitute the captured columns SHIFTED back under the
        # lambda (their own BoundRefs must re-shift by k)
        k = len(e.params)

        def shift(x: S.Expr) -> S.Expr:
            if isinstance(x, S.BoundRef):
                return S.BoundRef(x.index + k, x.name, x.dtype)
            ch = x.children()
            if not ch:
                return x
            out = x.with_children([shift(c) for c in ch])
            out.dtype = x.dtype
            return out

        shifted = [shift(sub.child if isinstance(sub, S.Alias) else sub)
                   for sub in exprs]
        params = [S.BoundRef(j, e.params[j], None) for j in range(k)]
        return S.Lambda(e.params, substitute_refs(e.body, params + shifted),
                        e.dtype)
    if isinstance(e, S.BoundRef):
        sub = exprs[e.index]
        return sub.child if isinstance(sub, S.Alias) else sub
    if isinstance(e, (S.ScalarSubquery, S.Exists)):
        return e
    if isinstance(e, S.InSubquery):
        return S.InSubquery(substitute_refs(e.child, exprs), e.plan, e.negated, e.dtype)
    if isinstance(e, S.Alias):
        return S.Alias(substitute_refs(e.child, exprs), e.name, e.dtype)
    ch = e.children()
    if not ch:
        return e
    out = e.with_children([substitute_refs(c, exprs) for c in ch])
    out.dtype = e.dtype
    return out


def split_conjuncts(e: S.Expr) -> List[S.Expr]:
    if isinstance(e, S.BinaryOp) and e.op == "and":
        return split_conjuncts(e.left) + split_conjuncts(e.right)
    return [e]


def conjoin(conds: List[S.Expr]) -> Optional[S.Expr]:
    if not conds:
        return None
    out = conds[0]
    for c in conds[1:]:
        out = S.BinaryOp("and", out, c, T.BOOL)
    return out


def expr_key(e: S.Expr) -> str:
    """Structural identity key (BoundRef indices included)."""
    if isinstance(e, S.BoundRef):
        return f"#{e.index}"
    if isinstance(e, S.Literal):
        return f"L{e.value!r}"
    if isinstance(e, S.BinaryOp):
        return f"({expr_key(e.left)}{e.op}{expr_key(e.right)})"
    if isinstance(e, S.UnaryOp):
        return f"{e.op}({expr_key(e.child)})"
    if isinstance(e, S.Cast):
        return f"cast({expr_key(e.child)},{e.to!r})"
    if isinstance(e, S.Func):
        return f"{e.name}({','.join(expr_key(a) for a in e.args)})"
    if isinstance(e, S.Alias):
        return expr_key(e.child)
    if isinstance(e, S.InList):
        return f"in({expr_key(e.child)},{[expr_key(v) for v in e.values]},{e.negated})"
    if isinstance(e, S.Between):
        return f"btw({expr_key(e.child)},{expr_key(e.low)},{expr_key(e.high)},{e.negated})"
    if isinstance(e, S.Like):
        return f"like({expr_key(e.child)},{expr_key(e.pattern)},{e.negated},{e.is_regex})"
    return repr(e)


def factor_common_disjuncts(e: S.Expr) -> S.Expr:
    """Rewrite OR(A∧X, A∧Y, ...) -> A ∧ OR(X, Y, ...): pulls join keys out of
    q19-style disjunctions so the equi-join extractor can see them."""
    if not (isinstance(e, S.BinaryOp) and e.op == "or"):
        return e
    disjuncts = _split_disjuncts(e)
    if len(disjuncts) < 2:
        return e
    conj_sets = [split_conjuncts(d) for d in disjuncts]
    keysets = [{expr_key(c) for c in cs} for cs in conj_sets]
    common_keys = set.intersection(*keysets)
    if not common_keys:
        return e
    common = []
    seen = set()
    for c in conj_sets[0]:
        k = expr_key(c)
        if k in common_keys and k not in seen:
            common.append(c)
            seen.add(k)
    rests = []
    for cs in conj_sets:
        rest = [c for c in cs if expr_key(c) not in common_keys]
        rests.append(conjoin(rest) or S.Literal(True, T.BOOL))
    ored = rests[0]
    for r in rests[1:]:
        ored = S.BinaryOp("or", ored, r, T.BOOL)
    return conjoin(common + [ored])


def _split_disjuncts(e: S.Expr) -> List[S.Expr]:
    if isinstance(e, S.BinaryOp) and e.op == "or":
        return _split_disjuncts(e.left) + _split_disjuncts(e.right)
    return [e]


def make_project(input_plan: S.Plan, indices: List[int]) -> S.Plan:
    """Projection selecting `indices` of input's schema."""
    exprs = [S.BoundRef(i, input_plan.schema[i][0], input_plan.schema[i][1]) for i in indices]
    p = S.Project(input=input_plan, exprs=exprs)
    p.schema = [input_plan.schema[i] for i in indices]
    return p
