"""Column pruning.

Narrows every operator to the columns its ancestors actually use. With
whole-partition columnar execution this is the rule that keeps giant string
columns (o_comment, p_name) from being gathered through joins they don't
participate in — the analogue of DataFusion's projection pushdown used by the
reference (ref: crates/sail-physical-optimizer/src/projection_pushdown.rs).

`_prune(plan, needed) -> (plan', mapping)` where mapping sends old output
ordinals to new ones; parents remap their expressions through it.
"""
from __future__ import annotations

from typing import Dict, List, Set, Tuple

from ...engine import types as T
from .. import spec as S
from .util import expr_refs, remap_expr


def prune_columns(plan: S.Plan) -> S.Plan:
    needed = set(range(len(plan.schema or [])))
    if not needed:
        # commands: prune their query inputs independently
        for attr in ("input",):
            child = getattr(plan, attr, None)
            if isinstance(child, S.Plan) and child.schema:
                setattr(plan, attr, prune_columns(child))
        return plan
    out, mapping = _prune(plan, needed)
    keep = sorted(needed)
    if len(out.schema) != len(keep) or any(mapping[i] != i for i in keep):
        # pass-through nodes (Filter/Sort) may carry extra predicate columns;
        # trim back to the original output shape
        pr = S.Project(input=out, exprs=[
            S.BoundRef(mapping[i], plan.schema[i][0], plan.schema[i][1]) for i in keep])
        pr.schema = [plan.schema[i] for i in keep]
        return pr
    return out


def _identity(plan: S.Plan) -> Tuple[S.Plan, Dict[int, int]]:
    return plan, {i: i for i in range(len(plan.schema))}


def _remap_subqueries(e: S.Expr) -> S.Expr:
    """Prune inside subquery plans (their own scopes)."""
    if isinstance(e, S.ScalarSubquery):
        return S.ScalarSubquery(plan=prune_columns(e.plan), dtype=e.dtype)
    if isinstance(e, S.Exists):
        return S.Exists(plan=prune_columns(e.plan), negated=e.negated, dtype=e.dtype)
    if isinstance(e, S.InSubquery):
        return S.InSubquery(_remap_subqueries(e.child), prune_columns(e.plan), e.negated, e.dtype)
    ch = e.children()
    if not ch:
        return e
    out = e.with_children([_remap_subqueries(c) for c in ch])
    out.dtype = e.dtype
    return out


def _prune(plan: S.Plan, needed: Set[int]) -> Tuple[S.Plan, Dict[int, int]]:
    # keep at least one column: a zero-column chunk would lose the row count
    # (count(*)-only aggregates, literal-only projections)
    if not needed and plan.schema:
        needed = {0}
    if plan.__dict__.get("_cte_cache_key") is not None:
        # shared CTE body: every use must see the identical (full) schema
        full = set(range(len(plan.schema)))
        if needed != full:
            key = plan.__dict__["_cte_cache_key"]
            plan.__dict__["_cte_cache_key"] = None
            out, mapping = _prune(plan, full)
            out.__dict__["_cte_cache_key"] = key
            plan.__dict__["_cte_cache_key"] = key
            return out, mapping
        key = plan.__dict__["_cte_cache_key"]
        plan.__dict__["_cte_cache_key"] = None
        out, mapping = _prune(plan, full)
        out.__dict__["_cte_cache_key"] = key
        plan.__dict__["_cte_cache_key"] = key
        return out, mapping
    if isinstance(plan, S.Project):
        keep = sorted(needed)
        kept_exprs = [_remap_subqueries(plan.exprs[i]) for i in keep]
        child_needed: Set[int] = set()
        for e in kept_exprs:
            child_needed |= expr_refs(e)
        child, cmap = _prune(plan.input, child_needed)
        new_exprs = [remap_expr(e, cmap) for e in kept_exprs]
        out = S.Project(input=child, exprs=new_exprs)
        out.schema = [plan.schema[i] for i in keep]
        return out, {old: new for new, old in enumerate(keep)}

    if isinstance(plan, S.Filter):
        child_needed = set(needed) | expr_refs(plan.condition)
        child, cmap = _prune(plan.input, child_needed)
        cond = remap_expr(_remap_subqueries(plan.condition), cmap)
        out = S.Filter(input=child, condition=cond)
        out.schema = child.schema
        if child_needed - set(needed):
            # predicate-only columns: trim IMMEDIATELY above the filter so
            # they are never gathered through joins/aggregates above (the
            # executor fuses Project∘Filter into a lazy selection)
            keep = sorted(needed)
            pr = S.Project(input=out, exprs=[
                S.BoundRef(cmap[i], out.schema[cmap[i]][0], out.schema[cmap[i]][1])
                for i in keep])
            pr.schema = [plan.schema[i] for i in keep]
            return pr, {old: new for new, old in enumerate(keep)}
        return out, cmap

    if isinstance(plan, S.SubqueryAlias):
        child, cmap = _prune(plan.input, needed)
        out = S.SubqueryAlias(input=child, alias=plan.alias)
        # keep this node's (possibly aliased) names for surviving columns
        out.schema = [(plan.schema[old][0] if old < len(plan.schema) else child.schema[new][0],
                       child.schema[new][1])
                      for old, new in sorted(cmap.items(), key=lambda kv: kv[1])]
        return out, cmap

    if isinstance(plan, S.Sort):
        child_needed = set(needed)
        for k in plan.keys:
            child_needed |= expr_refs(k)
        child, cmap = _prune(plan.input, child_needed)
        keys = [remap_expr(k, cmap) for k in plan.keys]
        out = S.Sort(input=child, keys=keys)
        out.schema = child.schema
        return out, cmap

    if isinstance(plan, S.Limit):
        child, cmap = _prune(plan.input, needed)
        out = S.Limit(input=child, n=plan.n, offset=plan.offset)
        out.schema = child.schema
        return out, cmap

    if isinstance(plan, S.Distinct):
        # distinct semantics depend on every column
        child, cmap = _prune(plan.input, set(range(len(plan.input.schema))))
        out = S.Distinct(input=child)
        out.schema = child.schema
        return out, cmap

    if isinstance(plan, S.Aggregate):
        ng = len(plan.group_by)
        keep_aggs = sorted(i - ng for i in needed if i >= ng)
        child_needed: Set[int] = set()
        for g in plan.group_by:
            child_needed |= expr_refs(g)
        kept_aggs = [_remap_subqueries(plan.aggs[i]) for i in keep_aggs]
        for a in kept_aggs:
            child_needed |= expr_refs(a)
        child, cmap = _prune(plan.input, child_needed)
        groups = [remap_expr(_remap_subqueries(g), cmap) for g in plan.group_by]
        aggs = [remap_expr(a, cmap) for a in kept_aggs]
        out = S.Aggregate(input=child, group_by=groups, aggs=aggs,
                          grouping_sets=plan.grouping_sets)
        out.schema = [plan.schema[i] for i in range(ng)] + \
                     [plan.schema[ng + i] for i in keep_aggs]
        mapping = {i: i for i in range(ng)}
        for new, old in enumerate(keep_aggs):
            mapping[ng + old] = ng + new
        return out, mapping

    if isinstance(plan, S.Join):
        nleft = len(plan.left.schema)
        on_refs = expr_refs(plan.on) if plan.on is not None else set()
        allrefs = set(needed) | on_refs
        lneeded = {i for i in allrefs if i < nleft}
        rneeded = {i - nleft for i in allrefs if i >= nleft}
        if plan.how in ("semi", "anti"):
            # output = left only; right side still needs its on-ref columns
            lneeded = {i for i in (set(needed) | {r for r in on_refs if r < nleft})}
            rneeded = {i - nleft for i in on_refs if i >= nleft}
        if not lneeded:
            lneeded = {0} if plan.left.schema else set()
        if not rneeded and plan.right.schema:
            rneeded = {0}
        left, lmap = _prune(plan.left, lneeded)
        right, rmap = _prune(plan.right, rneeded)
        nleft_new = len(left.schema)
        mapping = {}
        for old, new in lmap.items():
            mapping[old] = new
        for old, new in rmap.items():
            mapping[nleft + old] = nleft_new + new
        on = remap_expr(_remap_subqueries(plan.on), mapping) if plan.on is not None else None
        out = S.Join(left=left, right=right, how=plan.how, on=on, using=plan.using)
        if plan.how in ("semi", "anti"):
            out.schema = left.schema
            return out, lmap
        if plan.how in ("rightsemi", "rightanti"):
            out.schema = right.schema
            return out, rmap
        out.schema = list(left.schema) + list(right.schema)
        return out, mapping

    if isinstance(plan, S.SetOp):
        # positions must align: need the same set on both sides
        keep = sorted(needed)
        left, lmap = _prune_exact(plan.left, keep)
        right, rmap = _prune_exact(plan.right, keep)
        out = S.SetOp(op=plan.op, left=left, right=right, is_all=plan.is_all, by_name=plan.by_name)
        out.schema = [plan.schema[i] for i in keep]
        return out, {old: new for new, old in enumerate(keep)}

    if isinstance(plan, S.WindowPlan):
        nin = len(plan.input.schema)
        child_needed = {i for i in needed if i < nin}
        kept_w = sorted(i - nin for i in needed if i >= nin)
        wexprs = [plan.window_exprs[i] for i in kept_w]
        for w in wexprs:
            child_needed |= expr_refs(w)
            if isinstance(w, S.WindowExpr):
                for p in w.partition_by:
                    child_needed |= expr_refs(p)
                for k in w.order_by:
                    child_needed |= expr_refs(k)
                child_needed |= expr_refs(w.func)
        child, cmap = _prune(plan.input, child_needed)
        new_w = [remap_expr(w, cmap) for w in wexprs]
        out = S.WindowPlan(input=child, window_exprs=new_w)
        nin_new = len(child.schema)
        out.schema = list(child.schema) + [(plan.schema[nin + i][0], plan.schema[nin + i][1])
                                           for i in kept_w]
        mapping = dict(cmap)
        for new, old in enumerate(kept_w):
            mapping[nin + old] = nin_new + new
        return out, mapping

    if isinstance(plan, S.DataSourceRead):
        # push the column subset INTO the scan: file readers (parquet GPU
        # decode, pyarrow) only decode the schema's columns — the analogue
        # of DataFusion's scan projection pushdown
        keep = sorted(needed)
        if len(keep) == len(plan.schema):
            return _identity(plan)
        nd = S.DataSourceRead(format=plan.format, paths=plan.paths,
                              options=plan.options,
                              user_schema=plan.user_schema)
        nd.schema = [plan.schema[i] for i in keep]
        if "_table_name" in plan.__dict__:
            nd.__dict__["_table_name"] = plan.__dict__["_table_name"]
        return nd, {old: new for new, old in enumerate(keep)}

    if isinstance(plan, (S.Read, S.LocalRelation, S.Range)):
        keep = sorted(needed)
        if len(keep) == len(plan.schema):
            return _identity(plan)
        pr = S.Project(input=plan,
                       exprs=[S.BoundRef(i, plan.schema[i][0], plan.schema[i][1]) for i in keep])
        pr.schema = [plan.schema[i] for i in keep]
        return pr, {old: new for new, old in enumerate(keep)}

    # unknown node: keep everything
    return _identity(plan)


def _prune_exact(plan: S.Plan, keep: List[int]) -> Tuple[S.Plan, Dict[int, int]]:
    """Prune to exactly `keep` in order (SetOp sides need aligned positions)."""
    out, mapping = _prune(plan, set(keep))
    if [mapping[i] for i in keep] == list(range(len(keep))) and len(out.schema) == len(keep):
        return out, mapping
    pr = S.Project(input=out, exprs=[
        S.BoundRef(mapping[i], out.schema[mapping[i]][0], out.schema[mapping[i]][1]) for i in keep])
    pr.schema = [out.schema[mapping[i]] for i in keep]
    return pr, {old: new for new, old in enumerate(keep)}
