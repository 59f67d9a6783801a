"""DML execution: MERGE INTO / UPDATE / DELETE.

Columnar merge: matched pairs come from the equi-join machinery, a
cardinality check rejects targets matching multiple source rows
(ref: crates/sail-physical-plan/src/merge_cardinality_check.rs), and the
result table is assembled as [kept target rows] + [updated rows] +
[inserted rows] then committed back (catalog swap or Delta overwrite
commit — ref: crates/sail-delta-lake/src/logical/merge.rs flow).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..plan import spec as S
from . import types as T
from .chunk import Chunk
from .column import Column, StringColumn, Table
from .eval import broadcast
from .joins import equi_join


class MergeError(Exception):
    pass


def _load_target(executor, ref, schema) -> Chunk:
    kind, name = ref
    if kind == "catalog":
        t = executor.ctx.session.catalog.get_table_data(name, executor.ctx.device)
        if t is None:
            raise MergeError(f"no data for table {name}")
        return Chunk.from_table(t)
    from ..datasource.registry import read_source

    t = read_source(kind, [name], {}, schema, executor.ctx.device)
    return Chunk.from_table(t)


def _store_target(executor, ref, chunk: Chunk, schema):
    kind, name = ref
    if kind == "catalog":
        executor.ctx.session.catalog.register_table_chunk(name, chunk, schema)
    elif kind == "delta":
        from ..datasource.delta import replace_table

        replace_table(name, chunk)
    else:
        # copy-on-write rewrite for other table formats (iceberg)
        from ..datasource.registry import write_source

        write_source(kind, name, chunk, "overwrite", {}, None)


def _null_column(dtype: T.DataType, n: int, device) -> Column:
    from .eval import Scalar

    return broadcast(Scalar(None, dtype), n, device)


def execute_merge(executor, p: S.MergeInto) -> Chunk:
    ev = executor.ev
    tschema = p.__dict__["_target_schema"]
    tref = p.__dict__["_target_ref"]
    target = _load_target(executor, tref, tschema)
    source = executor._gather(executor.execute(p.source))
    nt, ns = target.num_rows, source.num_rows
    dev = target.device
    ntcols = len(target.columns)

    # equi keys from ON (target ordinals < ntcols)
    from .executor import split_join_condition

    equi, residual = split_join_condition(p.on, ntcols)
    if not equi:
        raise MergeError("MERGE requires at least one equality condition in ON")
    tkeys = [target.columns[i] for i, _ in equi]
    skeys = [source.columns[j - ntcols] for _, j in equi]
    t_idx, s_idx, counts = equi_join(skeys, tkeys, "inner")
    pair = Chunk([c.gather(t_idx) for c in target.columns]
                 + [c.gather(s_idx) for c in source.columns],
                 [n for n, _ in tschema] + list(source.names))
    if residual is not None:
        rmask = ev.eval_mask(residual, pair)
        t_idx, s_idx = t_idx[rmask], s_idx[rmask]
        pair = pair.filter_mask(rmask)
        counts = torch.zeros(nt, dtype=torch.int64, device=dev)
        counts.index_add_(0, t_idx, torch.ones(t_idx.shape[0], dtype=torch.int64, device=dev))
    if bool((counts > 1).any()):
        raise MergeError(
            "MERGE cardinality violation: a target row matched multiple source rows")

    # --- matched actions (first matching WHEN wins per row) ---------------
    npairs = pair.num_rows
    decided = torch.zeros(npairs, dtype=torch.bool, device=dev)
    delete_t = torch.zeros(nt, dtype=torch.bool, device=dev)
    update_parts: List[Chunk] = []
    for a in p.matched:
        amask = (ev.eval_mask(a.condition, pair) if a.condition is not None
                 else torch.ones(npairs, dtype=torch.bool, device=dev)) & ~decided
        decided |= amask
        if not bool(amask.any()):
            continue
        rows = torch.nonzero(amask, as_tuple=False).squeeze(1)
        if a.kind == "delete":
            delete_t[t_idx[rows]] = True
        else:  # update: build replacement rows from the pair scope
            sub = pair.gather(rows)
            delete_t[t_idx[rows]] = True  # original row replaced
            assigned = {n.lower(): e for n, e in a.assignments}
            cols = []
            for ci, (n, t) in enumerate(tschema):
                if n.lower() in assigned:
                    v = ev.eval(assigned[n.lower()], sub)
                    cols.append(broadcast(v, sub.num_rows, dev))
                else:
                    cols.append(sub.columns[ci])
            update_parts.append(Chunk(cols, [n for n, _ in tschema]))

    # --- not matched by source (target rows without any source match) -----
    unmatched_t = counts == 0
    for a in p.not_matched_by_source:
        scope_chunk = target
        amask = (ev.eval_mask(a.condition, scope_chunk) if a.condition is not None
                 else torch.ones(nt, dtype=torch.bool, device=dev)) & unmatched_t
        if a.kind == "delete":
            delete_t |= amask
        elif a.kind == "update" and bool(amask.any()):
            rows = torch.nonzero(amask, as_tuple=False).squeeze(1)
            sub = target.gather(rows)
            delete_t[rows] = True
            assigned = {n.lower(): e for n, e in a.assignments}
            cols = []
            for ci, (n, t) in enumerate(tschema):
                if n.lower() in assigned:
                    # bind scope is [target cols + source cols]; target-only
                    # refs are valid against `sub` directly
                    v = ev.eval(assigned[n.lower()], sub)
                    cols.append(broadcast(v, sub.num_rows, dev))
                else:
                    cols.append(sub.columns[ci])
            update_parts.append(Chunk(cols, [n for n, _ in tschema]))

    # --- not matched (inserts from source) ---------------------------------
    matched_s = torch.zeros(ns, dtype=torch.bool, device=dev)
    if s_idx.shape[0]:
        matched_s[s_idx] = True
    insert_parts: List[Chunk] = []
    s_decided = torch.zeros(ns, dtype=torch.bool, device=dev)
    for a in p.not_matched:
        # scope = [null target cols] + source
        scope_chunk = Chunk([_null_column(t, ns, dev) for _, t in tschema]
                            + list(source.columns),
                            [n for n, _ in tschema] + list(source.names))
        amask = (ev.eval_mask(a.condition, scope_chunk) if a.condition is not None
                 else torch.ones(ns, dtype=torch.bool, device=dev))
        amask = amask & ~matched_s & ~s_decided
        s_decided |= amask
        if not bool(amask.any()):
            continue
        rows = torch.nonzero(amask, as_tuple=False).squeeze(1)
        sub = scope_chunk.gather(rows)
        colmap = {c.lower(): v for c, v in zip(a.insert_columns, a.insert_values)}
        cols = []
        for n, t in tschema:
            if n.lower() in colmap:
                v = ev.eval(colmap[n.lower()], sub)
                col = broadcast(v, sub.num_rows, dev)
                col = _coerce_storage(col, t)
            else:
                col = _null_column(t, sub.num_rows, dev)
            cols.append(col)
        insert_parts.append(Chunk(cols, [n for n, _ in tschema]))

    # --- assemble ----------------------------------------------------------
    from .executor import concat_columns

    kept = target.filter_mask(~delete_t)
    parts = [kept] + update_parts + insert_parts
    parts = [c for c in parts if c.num_rows > 0] or [kept]
    out_cols = []
    for ci, (n, t) in enumerate(tschema):
        out_cols.append(concat_columns([_coerce_storage(c.columns[ci], t) for c in parts]))
    result = Chunk(out_cols, [n for n, _ in tschema])
    _store_target(executor, tref, result, tschema)
    n_updated = sum(c.num_rows for c in update_parts)
    n_inserted = sum(c.num_rows for c in insert_parts)
    n_deleted = int(delete_t.sum().item()) - n_updated
    return Chunk([Column.from_values([n_updated], T.I64, device="cpu"),
                  Column.from_values([max(n_deleted, 0)], T.I64, device="cpu"),
                  Column.from_values([n_inserted], T.I64, device="cpu")],
                 ["num_updated_rows", "num_deleted_rows", "num_inserted_rows"])


def _coerce_storage(col: Column, t: T.DataType) -> Column:
    from .eval import cast_column

    if col.dtype == t:
        return col
    return cast_column(col, t)


def execute_update(executor, p: S.UpdateTable) -> Chunk:
    ev = executor.ev
    tschema = p.__dict__["_target_schema"]
    tref = p.__dict__["_target_ref"]
    target = _load_target(executor, tref, tschema)
    dev = target.device
    mask = (ev.eval_mask(p.condition, target) if p.condition is not None
            else torch.ones(target.num_rows, dtype=torch.bool, device=dev))
    rows = torch.nonzero(mask, as_tuple=False).squeeze(1)
    sub = target.gather(rows)
    assigned = {n.lower(): e for n, e in p.assignments}
    cols = []
    for ci, (n, t) in enumerate(tschema):
        if n.lower() in assigned:
            v = ev.eval(assigned[n.lower()], sub)
            cols.append(_coerce_storage(broadcast(v, sub.num_rows, dev), t))
        else:
            cols.append(sub.columns[ci])
    updated = Chunk(cols, [n for n, _ in tschema])
    kept = target.filter_mask(~mask)
    from .executor import concat_columns

    out_cols = [concat_columns([kept.columns[i], updated.columns[i]])
                for i in range(len(tschema))]
    _store_target(executor, tref, Chunk(out_cols, [n for n, _ in tschema]), tschema)
    return Chunk([Column.from_values([int(rows.shape[0])], T.I64, device="cpu")],
                 ["num_affected_rows"])


#: DELETE rewrites data files only beyond this deleted fraction; below it
#: the commit is a deletion-vector update (no parquet rewrite)
DV_DELETE_MAX_FRACTION = 0.5


def execute_delete(executor, p: S.DeleteFrom) -> Chunk:
    ev = executor.ev
    tschema = p.__dict__["_target_schema"]
    tref = p.__dict__["_target_ref"]
    kind, name = tref
    if kind in ("delta", "iceberg"):
        # merge-on-read path: map deleted rows back to per-file positions
        # and commit deletion vectors (delta) / position-delete files
        # (iceberg) instead of rewriting parquet
        if kind == "delta":
            from ..datasource.delta import delete_with_dv as _commit_del
            from ..datasource.delta import scan_layout
        else:
            from ..datasource.iceberg import (delete_with_positions
                                              as _commit_del)
            from ..datasource.iceberg import scan_layout

        t, layout = scan_layout(name, tschema, executor.ctx.device, {})
        target = Chunk.from_table(t)
        dev = target.device
        mask = (ev.eval_mask(p.condition, target) if p.condition is not None
                else torch.ones(target.num_rows, dtype=torch.bool, device=dev))
        ndel = int(mask.sum().item())
        if ndel == 0:
            return Chunk([Column.from_values([0], T.I64, device="cpu")],
                         ["num_affected_rows"])
        if target.num_rows and ndel / target.num_rows <= DV_DELETE_MAX_FRACTION:
            _commit_del(name, layout, mask.cpu().numpy())
        else:
            _store_target(executor, tref, target.filter_mask(~mask), tschema)
        return Chunk([Column.from_values([ndel], T.I64, device="cpu")],
                     ["num_affected_rows"])
    target = _load_target(executor, tref, tschema)
    dev = target.device
    mask = (ev.eval_mask(p.condition, target) if p.condition is not None
            else torch.ones(target.num_rows, dtype=torch.bool, device=dev))
    kept = target.filter_mask(~mask)
    _store_target(executor, tref, kept, tschema)
    return Chunk([Column.from_values([int(mask.sum().item())], T.I64, device="cpu")],
                 ["num_affected_rows"])
