"""Thin-aggregate rewrite (late dimension materialization).

Pattern (q10's shape):

    Limit(Sort(Project(Aggregate(Join(..., dim, ...),
                                 group_by=[dim_pk, dim_payload..., ...]))))

Grouping by a dimension's primary key plus its payload columns forces the
join to gather every payload column (c_name/c_address/c_phone/c_comment are
raw strings gathered for 12M rows in q10) and the aggregate to hash them.
The payloads are functionally dependent on the PK, so:

  1. drop the payload columns from the grouping (column pruning then removes
     them from the join entirely),
  2. re-attach them ABOVE the Sort/Limit by re-joining a fresh copy of the
     dimension leaf on the PK — the payload gather then touches only the
     surviving rows (20 after q10's LIMIT).

PK detection is statistical: catalog ndv == the leaf's global row count
(the reference's join_reorder uses the same catalog-statistics machinery;
this rewrite has no direct reference analogue — it exists because whole-
partition string gathers are the measured GPU cost, see profiles/).
"""
from __future__ import annotations

import copy
from typing import Dict, List, Optional, Tuple

from ...engine import types as T
from .. import spec as S
from .join_order import _estimate, _leaf_column_source
from .util import expr_refs, remap_expr


def thin_aggregates(plan: S.Plan, stats) -> S.Plan:
    if stats is None:
        return plan
    return _walk(plan, stats)


def _walk(p: S.Plan, stats) -> S.Plan:
    out = _try_rewrite(p, stats)
    if out is not None:
        p = out  # rewritten top can't re-match (input is a Join, not an
        # Aggregate) and the rebuilt inner chain has no droppable payload
    for attr in ("input", "left", "right"):
        c = getattr(p, attr, None)
        if isinstance(c, S.Plan):
            setattr(p, attr, _walk(c, stats))
    return p


def _peel(p: S.Plan):
    while isinstance(p, S.SubqueryAlias):
        p = p.input
    return p


def _flatten_leaves(p: S.Plan, leaves: List[S.Plan], offs: List[int]):
    if isinstance(p, S.Join) and p.how in ("inner", "cross"):
        _flatten_leaves(p.left, leaves, offs)
        _flatten_leaves(p.right, leaves, offs)
        return
    leaves.append(p)
    offs.append((offs[-1] if offs else 0) + len(p.schema))


def _try_rewrite(node: S.Plan, stats) -> Optional[S.Plan]:
    # match [Limit] -> Sort -> Project -> Aggregate
    chain: List[S.Plan] = []
    p = node
    if isinstance(p, S.Limit):
        chain.append(p)
        p = _peel(p.input)
    if not isinstance(p, S.Sort):
        return None
    chain.append(p)
    p = _peel(p.input)
    if not isinstance(p, S.Project):
        return None
    proj = p
    agg = _peel(proj.input)
    if not isinstance(agg, S.Aggregate) or agg.grouping_sets or agg.having is not None:
        return None
    J = _peel(agg.input)
    mid = None  # optional Project between Aggregate and Join
    if isinstance(J, S.Project):
        mid = J
        J = _peel(mid.input)
    if not isinstance(J, S.Join) or J.how not in ("inner",):
        return None

    def to_join_coord(r: int) -> Optional[int]:
        if mid is None:
            return r
        e = mid.exprs[r]
        e = e.child if isinstance(e, S.Alias) else e
        return e.index if isinstance(e, S.BoundRef) else None

    # group keys must be plain column refs (traced through the mid Project)
    key_refs: List[Optional[int]] = []
    for g in agg.group_by:
        e = g.child if isinstance(g, S.Alias) else g
        if not isinstance(e, S.BoundRef):
            return None
        key_refs.append(to_join_coord(e.index))
    if any(r is None for r in key_refs):
        return None

    leaves: List[S.Plan] = []
    offs: List[int] = []
    _flatten_leaves(J, leaves, offs)
    starts = [0] + offs[:-1]

    def leaf_of(idx: int) -> Optional[int]:
        for li in range(len(leaves)):
            if starts[li] <= idx < starts[li] + len(leaves[li].schema):
                return li
        return None

    # aggregate-arg references (payloads must not feed any aggregate)
    agg_used = set()
    for a in agg.aggs:
        agg_used |= expr_refs(a)

    # find a leaf with a PK key + >=1 droppable payload keys
    by_leaf: Dict[int, List[int]] = {}
    for j, r in enumerate(key_refs):
        li = leaf_of(r)
        if li is None:
            return None
        by_leaf.setdefault(li, []).append(j)
    best = None
    for li, kjs in by_leaf.items():
        if len(kjs) < 2:
            continue
        est_rows = _estimate(leaves[li], stats)[1]
        for j in kjs:
            src = _leaf_column_source(leaves[li], key_refs[j] - starts[li])
            if src is None:
                continue
            cs = stats.column_stats(src[0], src[1]) if hasattr(stats, "column_stats") else None
            if cs is None or cs[1] is None:
                continue
            rows, ndv = cs
            t = agg.group_by[j].dtype
            if t is None or not getattr(t, "is_integer", False):
                continue
            if ndv >= max(rows, est_rows) * 0.99 and rows > 1000:
                payload = [k for k in kjs if k != j and key_refs[k] not in agg_used]
                # payload columns must be >cheap-to-rebuild (only worth it
                # when there is real payload to drop)
                if payload:
                    best = (li, j, payload)
                    break
        if best:
            break
    if best is None:
        return None
    li, det_j, payload_js = best
    payload_set = set(payload_js)
    nkeys = len(agg.group_by)

    # upper Project must use payload outputs only as passthrough, and Sort
    # keys must not touch them
    out_pos_of_key = {j: j for j in range(nkeys)}  # Aggregate schema: keys first
    payload_out = {out_pos_of_key[j] for j in payload_js}
    passthrough: Dict[int, int] = {}  # project expr index -> payload out pos
    for pi, e in enumerate(proj.exprs):
        inner = e.child if isinstance(e, S.Alias) else e
        refs = expr_refs(e)
        if refs & payload_out:
            if isinstance(inner, S.BoundRef) and inner.index in payload_out:
                passthrough[pi] = inner.index
            else:
                return None
    sort = next(c for c in chain if isinstance(c, S.Sort))
    for k in sort.keys:
        if expr_refs(k) & set(passthrough.keys()):
            # sort keys reference project outputs; compute which project
            # outputs are payload positions
            return None

    # ---- build the rewritten plan ----
    keep_keys = [j for j in range(nkeys) if j not in payload_set]
    new_out_of_old = {}
    for newj, oldj in enumerate(keep_keys):
        new_out_of_old[oldj] = newj
    nko = len(keep_keys)
    for ai in range(len(agg.aggs)):
        new_out_of_old[nkeys + ai] = nko + ai
    agg2 = S.Aggregate(input=agg.input,
                       group_by=[agg.group_by[j] for j in keep_keys],
                       aggs=agg.aggs)
    agg2.schema = [agg.schema[j] for j in keep_keys] + list(agg.schema[nkeys:])

    # inner Project without the payload passthroughs
    keep_pi = [pi for pi in range(len(proj.exprs)) if pi not in passthrough]
    proj2 = S.Project(input=agg2,
                      exprs=[remap_expr(proj.exprs[pi], new_out_of_old)
                             for pi in keep_pi])
    proj2.schema = [proj.schema[pi] for pi in keep_pi]
    out_of_pi = {pi: k for k, pi in enumerate(keep_pi)}

    sort2 = S.Sort(input=proj2,
                   keys=[remap_expr(k, out_of_pi) for k in sort.keys])
    sort2.schema = proj2.schema
    top: S.Plan = sort2
    lim = next((c for c in chain if isinstance(c, S.Limit)), None)
    if lim is not None:
        top = S.Limit(input=sort2, n=lim.n, offset=lim.offset)
        top.schema = sort2.schema

    # re-attach the dimension leaf on the PK
    leaf_copy = copy.deepcopy(leaves[li])
    det_out_pi = None
    for pi, e in enumerate(proj.exprs):
        inner = e.child if isinstance(e, S.Alias) else e
        if isinstance(inner, S.BoundRef) and inner.index == out_pos_of_key[det_j] \
                and pi in out_of_pi:
            det_out_pi = out_of_pi[pi]
            break
    if det_out_pi is None:
        return None
    det_leaf_col = key_refs[det_j] - starts[li]
    nleft = len(top.schema)
    cond = S.BinaryOp("=",
                      S.BoundRef(det_out_pi, top.schema[det_out_pi][0],
                                 top.schema[det_out_pi][1]),
                      S.BoundRef(nleft + det_leaf_col,
                                 leaf_copy.schema[det_leaf_col][0],
                                 leaf_copy.schema[det_leaf_col][1]), T.BOOL)
    rejoin = S.Join(left=top, right=leaf_copy, how="inner", on=cond)
    rejoin.schema = list(top.schema) + list(leaf_copy.schema)

    # final projection: original proj output order
    final_exprs: List[S.Expr] = []
    final_schema = []
    for pi in range(len(proj.exprs)):
        nm, t = proj.schema[pi]
        if pi in passthrough:
            old_out = passthrough[pi]
            oldj = old_out  # payload key output position == key index
            lc = key_refs[oldj] - starts[li]
            final_exprs.append(S.BoundRef(nleft + lc, nm, t))
        else:
            final_exprs.append(S.BoundRef(out_of_pi[pi], nm, t))
        final_schema.append((nm, t))
    out = S.Project(input=rejoin, exprs=final_exprs)
    out.schema = final_schema
    # re-apply the sort AFTER the rejoin (joins do not preserve order); the
    # final projection restores the ORIGINAL output positions, so the
    # original sort keys apply unchanged
    final_sort = S.Sort(input=out, keys=copy.deepcopy(sort.keys))
    final_sort.schema = out.schema
    return final_sort
