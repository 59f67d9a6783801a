"""Filter pushdown.

Pushes predicates toward the scans and converts comma-join cross products
into inner joins with ON conditions — the role of DataFusion's
FilterPushdown rule in the reference's physical optimizer stack
(ref: crates/sail-physical-optimizer/src/lib.rs FilterPushdown)."""
from __future__ import annotations

from typing import List, Optional

from ...engine import types as T
from .. import spec as S
from .util import (conjoin, expr_refs, factor_common_disjuncts, remap_expr,
                   split_conjuncts, substitute_refs)


def pushdown_filters(plan: S.Plan) -> S.Plan:
    return _push(plan, [])


def _has_subquery(e: S.Expr) -> bool:
    if isinstance(e, (S.ScalarSubquery, S.Exists, S.InSubquery)):
        return True
    return any(_has_subquery(c) for c in e.children())


def _is_volatile(e: S.Expr) -> bool:
    if isinstance(e, S.Func) and e.name in ("rand", "randn", "uuid", "monotonically_increasing_id"):
        return True
    return any(_is_volatile(c) for c in e.children())


def _push(plan: S.Plan, conds: List[S.Expr]) -> S.Plan:
    """Push the list of predicates (bound against `plan`'s schema) into the
    subtree; apply any that can't move as a Filter on the result."""
    key = plan.__dict__.get("_cte_cache_key")
    if key is not None:
        # shared CTE body: rewrites inside must not depend on the use site
        # (all uses share one executed result) and the cache marker must
        # survive node rebuilding
        plan.__dict__["_cte_cache_key"] = None
        out = _push(plan, [])
        plan.__dict__["_cte_cache_key"] = key
        out.__dict__["_cte_cache_key"] = key
        return _apply(out, conds)
    if isinstance(plan, S.Filter):
        newconds = []
        for c in split_conjuncts(plan.condition):
            newconds.append(factor_common_disjuncts(c))
        flat = []
        for c in newconds:
            flat.extend(split_conjuncts(c))
        return _push(plan.input, flat + conds)

    if isinstance(plan, S.Project):
        pushable, stuck = [], []
        for c in conds:
            if _has_subquery(c):
                stuck.append(c)
            else:
                pushable.append(substitute_refs(c, plan.exprs))
        inner = _push(plan.input, pushable)
        out = S.Project(input=inner, exprs=[_push_into_expr(e) for e in plan.exprs])
        out.schema = plan.schema
        return _apply(out, stuck)

    if isinstance(plan, S.SubqueryAlias):
        inner = _push(plan.input, conds)
        out = S.SubqueryAlias(input=inner, alias=plan.alias, column_aliases=plan.column_aliases)
        out.schema = plan.schema
        return out

    if isinstance(plan, (S.Sort, S.Limit, S.Distinct)):
        # limit: predicates cannot cross a limit boundary
        if isinstance(plan, S.Limit):
            inner = _push(plan.input, [])
            out = S.Limit(input=inner, n=plan.n, offset=plan.offset)
            out.schema = plan.schema
            return _apply(out, conds)
        inner = _push(plan.input, conds)
        if isinstance(plan, S.Sort):
            out = S.Sort(input=inner, keys=plan.keys)
        else:
            out = S.Distinct(input=inner)
        out.schema = plan.schema
        return out

    if isinstance(plan, S.Aggregate):
        ngroups = len(plan.group_by)
        pushable, stuck = [], []
        for c in conds:
            refs = expr_refs(c)
            if refs and all(i < ngroups for i in refs) and not _has_subquery(c):
                pushable.append(substitute_refs(c, plan.group_by))
            else:
                stuck.append(c)
        inner = _push(plan.input, pushable)
        out = S.Aggregate(input=inner, group_by=plan.group_by, aggs=plan.aggs,
                          grouping_sets=plan.grouping_sets)
        out.schema = plan.schema
        return _apply(out, stuck)

    if isinstance(plan, S.Join):
        return _push_join(plan, conds)

    if isinstance(plan, S.SetOp):
        if plan.op == "union":
            # same ordinal positions on both sides: push the predicates into each
            left = _push(plan.left, list(conds))
            right = _push(plan.right, list(conds))
            out = S.SetOp(op=plan.op, left=left, right=right, is_all=plan.is_all, by_name=plan.by_name)
            out.schema = plan.schema
            return out
        left = _push(plan.left, list(conds))
        right = _push(plan.right, [])
        out = S.SetOp(op=plan.op, left=left, right=right, is_all=plan.is_all, by_name=plan.by_name)
        out.schema = plan.schema
        return _apply(out, [])

    if isinstance(plan, S.WindowPlan):
        nin = len(plan.input.schema)
        pushable, stuck = [], []
        for c in conds:
            if all(i < nin for i in expr_refs(c)) and not _has_subquery(c):
                pushable.append(c)
            else:
                stuck.append(c)
        inner = _push(plan.input, pushable)
        out = S.WindowPlan(input=inner, window_exprs=plan.window_exprs)
        out.schema = plan.schema
        return _apply(out, stuck)

    if isinstance(plan, S.DataSourceRead) and plan.format == "parquet":
        # hive-partition pruning hint: equality conjuncts on scan columns are
        # COPIED (not moved) into scan options; the reader skips files whose
        # key=value path disagrees, the filter still applies for correctness
        for c in conds:
            if isinstance(c, S.BinaryOp) and c.op == "=":
                def _strip(x):
                    return x.child if isinstance(x, S.Cast) else x

                ref, lit = _strip(c.left), _strip(c.right)
                if isinstance(lit, S.BoundRef) and isinstance(ref, S.Literal):
                    ref, lit = lit, ref
                if isinstance(ref, S.BoundRef) and isinstance(lit, S.Literal) \
                        and not isinstance(lit.value, (list, dict)):
                    plan.options = dict(plan.options or {})
                    plan.options[f"partition.{ref.name}"] = str(lit.value)
        return _apply(plan, conds)

    # leaves and commands: recurse into children generically
    for attr in ("input",):
        if hasattr(plan, attr) and getattr(plan, attr) is not None and isinstance(getattr(plan, attr), S.Plan):
            setattr(plan, attr, _push(getattr(plan, attr), []))
    return _apply(plan, conds)


def _push_into_expr(e: S.Expr) -> S.Expr:
    """Recurse pushdown into subquery plans inside expressions."""
    if isinstance(e, S.ScalarSubquery):
        return S.ScalarSubquery(plan=_push(e.plan, []), dtype=e.dtype)
    if isinstance(e, S.Exists):
        return S.Exists(plan=_push(e.plan, []), negated=e.negated, dtype=e.dtype)
    if isinstance(e, S.InSubquery):
        return S.InSubquery(_push_into_expr(e.child), _push(e.plan, []), e.negated, e.dtype)
    ch = e.children()
    if not ch:
        return e
    out = e.with_children([_push_into_expr(c) for c in ch])
    out.dtype = e.dtype
    return out


def _push_join(plan: S.Join, conds: List[S.Expr]) -> S.Plan:
    nleft = len(plan.left.schema)
    ntotal = len(plan.left.schema) + len(plan.right.schema)
    how = plan.how

    join_conds = split_conjuncts(plan.on) if plan.on is not None else []
    all_conds = [factor_common_disjuncts(c) for c in conds]
    flat = []
    for c in all_conds:
        flat.extend(split_conjuncts(c))

    left_conds: List[S.Expr] = []
    right_conds: List[S.Expr] = []
    on_conds: List[S.Expr] = list(join_conds)
    post_conds: List[S.Expr] = []

    # semi/anti joins output only the left side: all filter refs are left refs
    left_only_output = how in ("semi", "anti")

    for c in flat:
        refs = expr_refs(c)
        if _has_subquery(c) or _is_volatile(c):
            post_conds.append(c)
            continue
        if left_only_output:
            left_conds.append(c)
            continue
        if all(i < nleft for i in refs):
            if how in ("inner", "left", "cross", "semi", "anti"):
                left_conds.append(c)
            else:  # right/full: left-side nulls possible -> stays post
                post_conds.append(c)
        elif all(i >= nleft for i in refs):
            if how in ("inner", "right", "cross"):
                right_conds.append(remap_expr(c, {i: i - nleft for i in refs}))
            else:
                post_conds.append(c)
        else:
            if how in ("inner", "cross"):
                on_conds.append(c)
            else:
                post_conds.append(c)

    # single-side ON conjuncts sink into their input where semantics allow:
    #  - inner/cross: both sides
    #  - left/semi/anti: right-side-only conjuncts only gate matching, so
    #    they can filter the right input (left rows survive as unmatched);
    #    left-side-only conjuncts must stay in ON for outer joins
    #  - right: mirror
    if how in ("inner", "cross", "left", "right", "semi", "anti"):
        sunk_on: List[S.Expr] = []
        for c in on_conds:
            refs = expr_refs(c)
            can_left = how in ("inner", "cross", "semi", "anti")
            can_right = how in ("inner", "cross", "left", "semi", "anti")
            if not _has_subquery(c) and refs and all(i < nleft for i in refs) and can_left:
                left_conds.append(c)
            elif not _has_subquery(c) and refs and all(i >= nleft for i in refs) and can_right:
                right_conds.append(remap_expr(c, {i: i - nleft for i in refs}))
            else:
                sunk_on.append(c)
        on_conds = sunk_on
        if on_conds and how == "cross":
            how = "inner"

    left = _push(plan.left, left_conds)
    right = _push(plan.right, right_conds)
    out = S.Join(left=left, right=right, how=how, on=conjoin(on_conds), using=plan.using)
    out.schema = plan.schema
    return _apply(out, post_conds)


def _apply(plan: S.Plan, conds: List[S.Expr]) -> S.Plan:
    # predicates that stay put may still carry subqueries: optimize inside them
    conds = [_push_into_expr(c) if _has_subquery(c) else c for c in conds]
    cond = conjoin(conds)
    if cond is None:
        return plan
    f = S.Filter(input=plan, condition=cond)
    f.schema = plan.schema
    return f
