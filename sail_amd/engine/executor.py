"""Plan executor.

Executes resolved (and optimized) logical plans over whole-partition Chunks.
Operator implementations here are the engine's reference semantics; on GPU
the hot operators (filter/project fusion, hash join, hash aggregate) are
routed through HIP kernels in ops/ with identical results.

The reference executes pull-based DataFusion streams per partition
(ref: crates/sail-execution/src/job_runner.rs:53 LocalJobRunner); here a
partition is the whole per-device table and operators run bottom-up.
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch

from ..plan import spec as S
from . import types as T
from .aggregates import agg_eval, global_ids, group_ids
from .chunk import Chunk
from .column import Column, StringColumn, Table
from .eval import Evaluator, Scalar, broadcast, cast_column, cast_value
from .joins import equi_join


from ..errors import ExecutionException


class ExecError(ExecutionException):
    pass


TOPK_MIN_ROWS = 1_000_000  # ORDER BY+LIMIT below this just sorts


class ExecutionContext:
    """Per-query execution context: session, device, subquery cache."""

    def __init__(self, session, device="cpu"):
        self.session = session
        self.device = torch.device(device)
        self._subquery_cache: Dict[int, Scalar] = {}
        self._cte_cache: Dict[int, Chunk] = {}

    def execute_scalar_subquery(self, e: S.ScalarSubquery) -> Scalar:
        if id(e) not in self._subquery_cache:
            chunk = Executor(self).execute(e.plan)
            if chunk.num_rows > 1:
                raise ExecError("scalar subquery returned more than one row")
            if chunk.num_rows == 0:
                val = Scalar(None, e.dtype)
            else:
                col = chunk.columns[0]
                v = col.to_pylist()[0]
                val = Scalar(v, e.dtype)
            self._subquery_cache[id(e)] = val
        return self._subquery_cache[id(e)]


class Executor:
    def __init__(self, ctx: ExecutionContext):
        self.ctx = ctx
        self.ev = Evaluator(ctx)
        d = getattr(ctx.session, "dist", None)
        self.dctx = d if (d is not None and d.world > 1) else None
        # tracer is shared across nested executors (subqueries) via the ctx
        if not hasattr(ctx, "tracer"):
            ctx.tracer = None
            import os

            if os.environ.get("SAIL_TRACE") == "1" or \
                    ctx.session.conf.get("sail.trace") == "true":
                from ..utils.trace import Tracer

                ctx.tracer = Tracer(ctx.device)
        self.tracer = ctx.tracer
        # §5.2 sanitizer analogue: serialize + error-check every operator so a
        # faulting kernel is attributed to the operator that launched it
        self._sync_kernels = (ctx.session.conf.get("sail.debug.sync_kernels")
                              == "true" and ctx.device.type == "cuda")

    def _gather(self, chunk: Chunk) -> Chunk:
        """Replicate a sharded chunk on every rank (all_gather over RCCL)."""
        if self.dctx is None or chunk.partitioning != "sharded":
            return chunk
        from ..exec.distributed import gather_chunk

        out = gather_chunk(chunk, self.dctx)
        out.partitioning = "replicated"
        return out

    def execute(self, plan: S.Plan) -> Chunk:
        ck = plan.__dict__.get("_cte_cache_key") if hasattr(plan, "__dict__") else None
        if ck is not None:
            cached = self.ctx._cte_cache.get(ck)
            if cached is not None:
                return cached
        m = getattr(self, "_x_" + type(plan).__name__, None)
        if m is None:
            raise ExecError(f"cannot execute {type(plan).__name__}")
        if self.tracer is not None:
            detail = ""
            if isinstance(plan, S.Read):
                detail = plan.table
            elif isinstance(plan, S.Join):
                detail = plan.how
            elif isinstance(plan, S.Aggregate):
                detail = f"{len(plan.group_by)}keys/{len(plan.aggs)}aggs"
            out = self.tracer.wrap(type(plan).__name__, detail, lambda: m(plan))
        else:
            out = m(plan)
        if ck is not None:
            self.ctx._cte_cache[ck] = out
        if self._sync_kernels:
            try:
                torch.cuda.synchronize(self.ctx.device)
            except RuntimeError as e:
                raise ExecError(
                    f"device fault inside {type(plan).__name__}: {e}") from e
        return out

    # -- leaves ------------------------------------------------------------
    def _x_Read(self, p: S.Read) -> Chunk:
        cat = self.ctx.session.catalog
        t = cat.get_table_data(p.table, self.ctx.device)
        if t is None:
            raise ExecError(f"no data for table {p.table}")
        out = Chunk.from_table(t)
        if self.dctx is not None and not cat.is_replicated(p.table):
            out.partitioning = "sharded"
        return out

    def _x_TableFuncRead(self, p: S.TableFuncRead) -> Chunk:
        info = self.ctx.session.udtfs.get(p.name.lower())
        if info is None:
            raise ExecError(f"table function {p.name} not registered")
        fn, schema = info
        argvals = []
        for a in p.args:
            v = self.ev.eval(a, Chunk([], [], forced_rows=1))
            argvals.append(v.value if isinstance(v, Scalar) else v.to_pylist()[0])
        data = fn(*argvals)
        cols = [Column.from_values(data[n], t, device=self.ctx.device)
                for n, t in schema]
        return Chunk(cols, [n for n, _ in schema])

    def _x_ChunkSource(self, p: S.ChunkSource) -> Chunk:
        return p.chunk

    def _x_DataSourceRead(self, p: S.DataSourceRead) -> Chunk:
        from ..datasource.registry import read_source

        t = read_source(p.format, p.paths, p.options, p.schema, self.ctx.device)
        if p.schema:
            want = [n for n, _ in p.schema]
            have = list(t.columns.keys())
            if have != want and all(n in t.columns for n in want):
                # readers that ignore the pruned schema still return the
                # full table; trim to the scan's declared columns
                t = t.select(want)
        out = Chunk.from_table(t)
        if (p.options or {}).get("partitioning") == "sharded" \
                and self.dctx is not None:
            out.partitioning = "sharded"
        return out

    def _x_LocalRelation(self, p: S.LocalRelation) -> Chunk:
        cols = []
        names = []
        for (name, dtype) in p.schema:
            cols.append(Column.from_values(p.data[name], dtype, device=self.ctx.device))
            names.append(name)
        return Chunk(cols, names)

    def _x_Range(self, p: S.Range) -> Chunk:
        data = torch.arange(p.start, p.end, p.step, dtype=torch.int64, device=self.ctx.device)
        return Chunk([Column(T.I64, data)], ["id"])

    def _x_SubqueryAlias(self, p: S.SubqueryAlias) -> Chunk:
        c = self.execute(p.input)
        return Chunk(c.columns, [n for n, _ in p.schema], c.partitioning)

    # -- row ops -----------------------------------------------------------
    def _x_Filter(self, p: S.Filter) -> Chunk:
        child = self.execute(p.input)
        if child.num_rows == 0:
            return child
        mask = self.ev.eval_mask(p.condition, child)
        return child.filter_mask(mask)

    def _x_Project(self, p: S.Project) -> Chunk:
        # Project∘Filter fusion: evaluate the predicate on the unfiltered
        # child, then gather ONLY the columns the projection references —
        # predicate-only columns (e.g. o_comment in q13) are never gathered.
        fin = p.input
        while isinstance(fin, S.SubqueryAlias):
            fin = fin.input
        if isinstance(fin, S.Filter):
            f = fin
            base = self.execute(f.input)
            if base.num_rows == 0:
                child = base
            else:
                mask = self.ev.eval_mask(f.condition, base)
                idx = torch.nonzero(mask, as_tuple=False).squeeze(1)
                from ..plan.rules.util import expr_refs

                refs = set()
                for e in p.exprs:
                    refs |= expr_refs(e)
                cols: List[Optional[Column]] = [None] * len(base.columns)
                for i in refs:
                    cols[i] = base.columns[i].gather(idx)
                child = Chunk(cols, list(base.names), base.partitioning)
                child.forced_rows = int(idx.shape[0])
        else:
            child = self.execute(p.input)
        cols = []
        for e in p.exprs:
            v = self.ev.eval(e, child)
            cols.append(broadcast(v, child.num_rows, child.device))
        return Chunk(cols, [n for n, _ in p.schema], child.partitioning)

    def _x_Limit(self, p: S.Limit) -> Chunk:
        fin = p.input
        while isinstance(fin, S.SubqueryAlias):
            fin = fin.input
        if isinstance(fin, S.Sort) and p.n is not None \
                and (p.n + p.offset) <= 100_000:
            out = self._try_topk(fin, p.n + p.offset)
            if out is not None:
                start = p.offset
                return out.slice(start, max(0, min(p.n, out.num_rows - start)))
        child = self._gather(self.execute(p.input))
        start = p.offset
        n = p.n if p.n is not None else child.num_rows - start
        return child.slice(start, max(0, min(n, child.num_rows - start)))

    def _try_topk(self, sort: S.Sort, k: int) -> Optional[Chunk]:
        """ORDER BY ... LIMIT k over a big input: torch.topk SELECTION on the
        primary key (+ boundary ties), then the full stable multi-key sort on
        that candidate set only — exact, and skips sorting 90M rows for a
        top-10 (ClickBench q32-35 sorted the whole group table; profiles/
        showed the onesweep sorts as the dominant kernels)."""
        child = self._gather(self.execute(sort.input))
        N = child.num_rows
        if N < TOPK_MIN_ROWS or N <= 4 * k or not sort.keys:
            if N <= 1:
                return child
            return child.gather(sort_indices(self.ev, sort.keys, child))
        k0 = sort.keys[0]
        col = broadcast(self.ev.eval(k0.child, child), N, child.device)
        keyvals = _sortable(col)
        nulls_first = k0.nulls_first if k0.nulls_first is not None else k0.ascending
        if col.validity is not None:
            big = _null_sentinel(keyvals, nulls_first == k0.ascending)
            keyvals = torch.where(col.valid_mask(), keyvals, big)
        kc = min(N, max(4 * k, k + 1024))
        top = torch.topk(keyvals.to(torch.float64)
                         if keyvals.dtype not in (torch.int64, torch.float64)
                         else keyvals, kc, largest=not k0.ascending, sorted=True)
        boundary = top.values[k - 1]
        ties = torch.nonzero(keyvals == boundary.to(keyvals.dtype),
                             as_tuple=False).flatten()
        if ties.numel() > 5_000_000:
            return child.gather(sort_indices(self.ev, sort.keys, child))
        cand = torch.unique(torch.cat([top.indices, ties]))
        sub = child.gather(cand)
        order = sort_indices(self.ev, sort.keys, sub)
        return sub.gather(order[:k])

    DIST_DISTINCT_MIN_ROWS = int(os.environ.get(
        "SAIL_DIST_DISTINCT_MIN_ROWS", "2000000"))

    def _x_Distinct(self, p: S.Distinct) -> Chunk:
        child = self.execute(p.input)
        if self.dctx is not None and child.partitioning == "sharded":
            # local distinct first (shrinks the exchange)
            if child.num_rows:
                gid, rep, ng = group_ids(child.columns)
                child = Chunk([c.gather(rep) for c in child.columns],
                              list(child.names), "sharded")
            total = self.dctx.consensus_sum(child.num_rows)
            if total >= self.DIST_DISTINCT_MIN_ROWS:
                # big survivor set: hash-shuffle so each value lands on one
                # rank, dedup locally, stay sharded (VERDICT r1 item 2 —
                # no whole-table gather)
                from ..exec.distributed import shuffle_chunk

                child = shuffle_chunk(child, list(range(len(child.columns))),
                                      self.dctx)
                if child.num_rows == 0:
                    return child
                gid, rep, ng = group_ids(child.columns)
                return Chunk([c.gather(rep) for c in child.columns],
                             list(child.names), "sharded")
            child = self._gather(child)
        if child.num_rows == 0:
            return child
        gid, rep, ng = group_ids(child.columns)
        return child.gather(rep)

    def _x_RecursionRef(self, p: S.RecursionRef) -> Chunk:
        frames = getattr(self.ctx, "recursion_frames", None)
        if not frames or p.name not in frames:
            raise ExecError(f"recursion reference {p.name} outside recursive CTE")
        return frames[p.name]

    def _x_RecursiveCte(self, p: S.RecursiveCte) -> Chunk:
        """Fixpoint iteration: result += recursive(working_set) until empty
        (ref: Spark recursive CTE semantics; sail-plan recursion.rs).
        UNION (distinct) deduplicates each delta against everything seen —
        required for cyclic-graph termination."""
        result = self.execute(p.anchor)
        if not hasattr(self.ctx, "recursion_frames"):
            self.ctx.recursion_frames = {}
        frames = self.ctx.recursion_frames
        if not p.is_all:
            result = self._dedup_chunk(result)
        seen_keys = self._row_keys(result) if not p.is_all else None
        work = result
        names = [nm for nm, _ in p.schema]
        it = 0
        while work.num_rows > 0:
            it += 1
            if it > p.max_iter:
                raise ExecError(
                    f"recursive CTE {p.name}: exceeded {p.max_iter} iterations "
                    "(set sail.execution.max_recursion to raise)")
            frames[p.name] = work
            self.ctx._cte_cache.clear()  # the body must re-execute per iteration
            new = self.execute(p.recursive)
            if not p.is_all and new.num_rows:
                keys = self._row_keys(new)
                srt = torch.sort(seen_keys).values
                pos = torch.searchsorted(srt, keys).clamp_max(max(srt.shape[0] - 1, 0))
                fresh_mask = ~((srt.shape[0] > 0) & (srt.index_select(0, pos) == keys))
                idx = torch.nonzero(fresh_mask, as_tuple=False).flatten()
                new = Chunk([c.gather(idx) for c in new.columns], list(new.names))
                new = self._dedup_chunk(new)
                if new.num_rows:
                    seen_keys = torch.cat([seen_keys, self._row_keys(new)])
            if new.num_rows == 0:
                break
            result = Chunk([concat_columns([a, b]) for a, b in
                            zip(result.columns, new.columns)], names)
            work = new
        frames.pop(p.name, None)
        result.names = names
        return result

    def _row_keys(self, chunk: Chunk) -> torch.Tensor:
        """Cross-chunk-stable row keys (NOT _pack_or_hash, whose codes are
        chunk-local): raw values for ints, FNV for strings, mix64-combined.
        Hash-based — the engine-wide n²/2⁶⁴ collision tradeoff."""
        from ..exec.distributed import _mix64
        from .joins import fnv_key_tensor, normalize_key

        if not chunk.columns:
            return torch.zeros(chunk.num_rows, dtype=torch.int64)
        acc = None
        for c in chunk.columns:
            if isinstance(c, StringColumn):
                k = fnv_key_tensor(c.decode_dict())
            else:
                k = normalize_key(c)
            # fold nulls to a distinct sentinel so NULL == NULL for dedup
            if c.validity is not None:
                k = torch.where(c.valid_mask(), k,
                                torch.full_like(k, -(1 << 61) + 3))
            acc = k if acc is None else _mix64(acc * 31 + k)
        return acc

    def _dedup_chunk(self, chunk: Chunk) -> Chunk:
        if chunk.num_rows <= 1:
            return chunk
        keys = self._row_keys(chunk)
        order = torch.argsort(keys, stable=True)
        sk = keys.index_select(0, order)
        first = torch.ones(sk.shape[0], dtype=torch.bool, device=sk.device)
        first[1:] = sk[1:] != sk[:-1]
        kept = order[first].sort().values
        return Chunk([c.gather(kept) for c in chunk.columns], list(chunk.names),
                     chunk.partitioning)

    def _x_Generate(self, p: S.Generate) -> Chunk:
        """explode/posexplode: repeat parent rows per array element
        (repeat_interleave over the list offsets — one vectorized pass,
        no per-row loop)."""
        from .column import ListColumn, MapColumn, StructColumn

        child = self.execute(p.input)
        n, dev = child.num_rows, child.device
        gen = broadcast(self.ev.eval(p.gen, child), n, dev)
        is_map = isinstance(gen, MapColumn)
        if not isinstance(gen, (ListColumn, MapColumn)):
            raise ExecError("generator input is not an array or map")
        lens = (gen.offsets[1:] - gen.offsets[:-1]) if is_map \
            else gen.lengths()
        valid = gen.valid_mask()
        if p.outer:
            eff = torch.where(valid & (lens > 0), lens, torch.ones_like(lens))
            has_elem = valid & (lens > 0)
        else:
            eff = torch.where(valid, lens, torch.zeros_like(lens))
            has_elem = None  # every emitted row has a real element
        offs2 = torch.zeros(n + 1, dtype=torch.int64, device=dev)
        torch.cumsum(eff, 0, out=offs2[1:])
        total = int(offs2[-1].item())
        parent = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev), eff)
        pos = torch.arange(total, dtype=torch.int64, device=dev) \
            - torch.repeat_interleave(offs2[:-1], eff)
        child_idx = gen.offsets[:-1].index_select(0, parent) + pos
        srcs = [gen.keys, gen.values] if is_map else [gen.child]
        clen = len(srcs[0])
        child_idx = child_idx.clamp(0, max(clen - 1, 0))
        elems = []
        for src in srcs:
            e = src.gather(child_idx) if clen else \
                Column.from_values([None] * total, src.dtype, device=dev)
            if has_elem is not None:
                ev_mask = has_elem.index_select(0, parent) & e.valid_mask()
                e.validity = None if bool(ev_mask.all()) \
                    else ev_mask.to(torch.uint8)
            elems.append(e)
        if getattr(p, "mode", "") == "inline" and len(elems) == 1 \
                and isinstance(elems[0], StructColumn):
            st = elems[0]
            sval = st.valid_mask()
            expanded = []
            for _nm, fc in st.children_:
                fv = fc.valid_mask() & sval
                fc = fc.gather(torch.arange(len(fc), device=dev))
                fc.validity = None if bool(fv.all()) else fv.to(torch.uint8)
                expanded.append(fc)
            elems = expanded
        cols = [c.gather(parent) for c in child.columns]
        if p.position:
            cols.append(Column(T.I32, pos.to(torch.int32)))
        cols.extend(elems)
        out = Chunk(cols, [nm for nm, _ in p.schema], child.partitioning)
        return out

    def _x_Sample(self, p: S.Sample) -> Chunk:
        """Bernoulli sampling with a deterministic generator (REPEATABLE
        seed, or a fixed default so every SPMD rank draws the same mask for
        replicated chunks); n ROWS takes the first n (Spark semantics)."""
        child = self.execute(p.input)
        if p.rows is not None:
            if child.num_rows <= p.rows:
                return child
            return child.gather(torch.arange(p.rows, dtype=torch.int64,
                                             device=child.device))
        n = child.num_rows
        if n == 0 or p.fraction is None:
            return child
        g = torch.Generator(device="cpu")
        g.manual_seed(p.seed if p.seed is not None else 0x5A11)
        mask = torch.rand(n, generator=g) < p.fraction
        idx = torch.nonzero(mask, as_tuple=False).flatten().to(child.device)
        return child.gather(idx)

    # -- sort --------------------------------------------------------------
    #: global-row thresholds below which gather+local beats a shuffle
    #: (env-overridable so the world=2 CPU tests can force the shuffle path)
    DIST_SORT_MIN_ROWS = int(os.environ.get("SAIL_DIST_SORT_MIN_ROWS",
                                            "2000000"))

    def _x_Sort(self, p: S.Sort) -> Chunk:
        child = self.execute(p.input)
        if (self.dctx is not None and child.partitioning == "sharded"
                and p.keys
                # decision must be rank-consistent: use the GLOBAL row count
                and self.dctx.consensus_sum(child.num_rows)
                >= self.DIST_SORT_MIN_ROWS):
            out = self._dist_sort(p, child)
            if out is not None:
                return out
        child = self._gather(child)
        if child.num_rows <= 1:
            return child
        idx = sort_indices(self.ev, p.keys, child)
        return child.gather(idx)

    def _dist_sort(self, p: S.Sort, child: Chunk) -> Optional[Chunk]:
        """Range-partitioned distributed sort: sampled
        splitters -> all_to_all by key range -> local multi-key sort ->
        rank-order all_gather (ranks own disjoint primary-key ranges, so
        the concat IS the global order). The O(N)-per-rank sort workspace
        becomes O(N/world); only the final replication is whole-table.
        None => caller takes the gather path (rank-local string ranks are
        not cross-rank comparable)."""
        from ..exec.distributed import (gather_chunk, range_partition_ids,
                                        shuffle_chunk_by_pids)

        k0 = p.keys[0]
        n = child.num_rows
        col = broadcast(self.ev.eval(k0.child, child), n, child.device)
        if isinstance(col, StringColumn):
            return None  # _sortable ranks are rank-local for strings
        keyvals = _sortable(col)
        nulls_first = k0.nulls_first if k0.nulls_first is not None \
            else k0.ascending
        if col.validity is not None:
            big = _null_sentinel(keyvals, nulls_first == k0.ascending)
            keyvals = torch.where(col.valid_mask(), keyvals, big)
        pids = range_partition_ids(keyvals, self.dctx,
                                   descending=not k0.ascending)
        local = shuffle_chunk_by_pids(child, pids, self.dctx)
        if local.num_rows > 1:
            idx = sort_indices(self.ev, p.keys, local)
            local = local.gather(idx)
        out = gather_chunk(local, self.dctx)
        out.partitioning = "replicated"
        return out

    # -- aggregate ---------------------------------------------------------
    #: scans above this many file bytes stream through row-group batches
    #: with a partial/merge aggregation state instead of materializing the
    #: whole table in HBM (out-of-core; ref: the
    #: reference's memory pools + spill, application.yaml:21-67)
    STREAM_SCAN_BYTES = int(os.environ.get("SAIL_EXEC_STREAM_SCAN_BYTES",
                                           str(150 << 30)))
    STREAM_SCAN_BATCH_ROWS = int(os.environ.get(
        "SAIL_EXEC_STREAM_SCAN_BATCH_ROWS", "16000000"))

    def _x_Aggregate(self, p: S.Aggregate) -> Chunk:
        if p.grouping_sets:
            return self._grouping_sets_aggregate(p)
        streamed = self._try_streamed_scan_aggregate(p)
        if streamed is not None:
            return streamed
        # Aggregate∘[Project]∘Filter fusion: evaluate group keys and agg
        # inputs on the UNFILTERED child and pass the selection mask into the
        # aggregation — avoids materializing high-selectivity filters (Q1
        # keeps 98.6% of lineitem; the gather costs more than the aggregate).
        fused = self._try_masked_aggregate(p) \
            if not p.grouping_sets and self._find_session_window(p) is None \
            else None
        if fused is not None:
            return fused
        child = self.execute(p.input)
        sw = self._find_session_window(p)
        if self.dctx is not None and child.partitioning == "sharded":
            if sw is None:
                return self._dist_aggregate(p, child)
            child = self._gather(child)  # sessions span shard boundaries
        n = child.num_rows
        dev = child.device
        if p.group_by:
            key_cols = [self._session_window_col(sw[1], j, p, child, n, dev)
                        if sw is not None and j == sw[0]
                        else broadcast(self.ev.eval(g, child), n, dev)
                        for j, g in enumerate(p.group_by)]
            if n == 0:
                return Chunk(key_cols + [_empty_agg_col(a, dev) for a in p.aggs],
                             [nm for nm, _ in p.schema])
            args_list = [[broadcast(self.ev.eval(x, child), n, dev) for x in a.args]
                         for a in p.aggs]
            fmasks = [self.ev.eval_mask(a.filter, child) if a.filter is not None else None
                      for a in p.aggs]
            from .aggregates import fused_agg_batch, try_direct_aggregate

            d = try_direct_aggregate(key_cols, p.aggs, args_list, fmasks)
            if d is not None:
                out_keys, agg_cols = d
                return Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema])
            gid, rep, ng = group_ids(key_cols)
            out_keys = [c.gather(rep) for c in key_cols]
        else:
            if n == 0:
                # global aggregate over empty input still yields one row
                cols = [_empty_global_agg(a, dev) for a in p.aggs]
                return Chunk(cols, [nm for nm, _ in p.schema])
            gid, ng = global_ids(n, dev)
            out_keys = []
            args_list = [[broadcast(self.ev.eval(x, child), n, dev) for x in a.args]
                         for a in p.aggs]
            fmasks = [self.ev.eval_mask(a.filter, child) if a.filter is not None else None
                      for a in p.aggs]
            from .aggregates import fused_agg_batch

        from .aggregates import fused_agg_batch

        agg_cols = fused_agg_batch(p.aggs, args_list, fmasks, gid, ng)
        if agg_cols is None:
            agg_cols = [agg_eval(a.name, args, gid, ng, a.distinct, fmask, a.dtype)
                        for a, args, fmask in zip(p.aggs, args_list, fmasks)]
        return Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema])

    def _try_streamed_scan_aggregate(self, p: S.Aggregate) -> Optional[Chunk]:
        """Aggregate over a parquet scan whose bytes exceed the streaming
        budget: decode row-group batches, fold each into a partial/merge
        aggregation state (streaming's _AggState — the identical
        decomposition), finalize once. Bounded memory: one batch + the
        group state resident at a time."""
        if self.dctx is not None:  # SPMD path merges partials its own way
            return None
        node = p.input
        chain = []
        while isinstance(node, (S.Project, S.Filter, S.SubqueryAlias)):
            if node.__dict__.get("_cte_cache_key") is not None:
                return None  # shared subtree: caching assumes one execution
            chain.append(node)
            node = node.input
        import glob as _g
        import os as _os

        def _scan_files(nd):
            out = []
            for pth in nd.paths or []:
                if _os.path.isdir(pth):
                    out += _g.glob(_os.path.join(pth, "**", "*.parquet"),
                                   recursive=True)
                elif _os.path.isfile(pth):
                    out.append(pth)
            return out, sum(_os.path.getsize(f) for f in out)

        join = None
        join_side = None
        if isinstance(node, S.Join):
            # Aggregate over (big scan ⋈ small side): stream the scan side
            # through the join+aggregation, build the other side ONCE.
            # Decomposable only when every streamed batch's matches are
            # independent: inner/cross always; left/semi/anti only when the
            # STREAMED side is the preserved (left) side; mirrored for
            # right joins (ref: out-of-core probe-streamed hash join).
            join = node
            for side in ("left", "right"):
                sub = getattr(join, side)
                c2, nd = [], sub
                while isinstance(nd, (S.Project, S.Filter, S.SubqueryAlias)):
                    if nd.__dict__.get("_cte_cache_key") is not None:
                        nd = None
                        break
                    c2.append(nd)
                    nd = nd.input
                if isinstance(nd, S.DataSourceRead) and \
                        nd.format == "parquet" and \
                        nd.__dict__.get("_cte_cache_key") is None and \
                        _scan_files(nd)[1] >= self.STREAM_SCAN_BYTES:
                    how = (join.how or "inner").lower()
                    ok = how in ("inner", "cross") or \
                        (side == "left" and how in
                         ("left", "left_outer", "semi", "left_semi",
                          "anti", "left_anti")) or \
                        (side == "right" and how in ("right", "right_outer"))
                    if ok:
                        join_side, chain, node = side, chain + [join] + c2, nd
                        break
            if join_side is None:
                return None
        if not isinstance(node, S.DataSourceRead) or node.format != "parquet":
            return None
        if node.__dict__.get("_cte_cache_key") is not None:
            return None
        files, total_bytes = _scan_files(node)
        if not files or total_bytes < self.STREAM_SCAN_BYTES:
            return None
        from ..exec.distributed import decompose_agg

        decomps = [decompose_agg(a) for a in p.aggs]
        if any(d is None for d in decomps) or \
                any(getattr(a, "distinct", False) for a in p.aggs):
            return None
        from ..datasource.parquet_io import scan_batches
        from ..streaming.query import _AggState

        st = _AggState(p, decomps)
        parent = chain[-1] if chain else None
        other_side = "right" if join_side == "left" else "left"
        orig_other = None
        if join is not None:
            # build the small side once; every streamed batch re-joins
            # against this resident chunk
            built = self.execute(getattr(join, other_side))
            orig_other = getattr(join, other_side)
            setattr(join, other_side, S.ChunkSource(
                chunk=built, schema=orig_other.schema))
        orig_input = (getattr(parent, "input", None)
                      if parent is not None and parent is not join else None)
        orig_stream = getattr(join, join_side) if join is not None else None
        finalized = None

        def _splice(src):
            if parent is None:
                return src
            if parent is join:
                setattr(join, join_side, src)
            else:
                parent.input = src
            return p.input

        try:
            for table in scan_batches(node.paths, node.schema, self.ctx.device,
                                      node.options,
                                      target_rows=self.STREAM_SCAN_BATCH_ROWS):
                src = S.ChunkSource(chunk=Chunk.from_table(table),
                                    schema=node.schema)
                child = self.execute(_splice(src))
                finalized, _ = st.update(self, child)
        finally:
            if parent is not None and parent is not join:
                parent.input = orig_input
            if join is not None:
                setattr(join, other_side, orig_other)
                setattr(join, join_side, orig_stream)
        if finalized is None:  # zero batches: fall back to the normal path
            return None
        return finalized

    @staticmethod
    def _find_session_window(p: S.Aggregate):
        """(index, Func) of a session_window(ts, gap) group key, or None."""
        for i, g in enumerate(p.group_by or []):
            e = g.child if isinstance(g, S.Alias) else g
            if isinstance(e, S.Func) and e.name.lower() == "session_window":
                return i, e
        return None

    def _session_window_col(self, e: S.Func, idx: int, p: S.Aggregate,
                            child: Chunk, n: int, dev) -> Column:
        """session_window(ts, gap): gap-separated sessions computed per
        sibling group-key partition — rows whose event times are within
        `gap` of the previous row (same partition) share a session; the
        struct is (min_ts, max_ts + gap) of the session (Spark semantics;
        ref: Spark session windows / sail streaming rewriter)."""
        from .column import StructColumn
        from .eval import Scalar
        from .functions_impl import _parse_duration_us

        ts = broadcast(self.ev.eval(e.args[0], child), n, dev)
        gv = self.ev.eval(e.args[1], child)
        gap_str = gv.value if isinstance(gv, Scalar) else gv.to_pylist()[0]
        gap = _parse_duration_us(gap_str) if isinstance(gap_str, str) \
            else int(gap_str)
        others = [broadcast(self.ev.eval(g, child), n, dev)
                  for j, g in enumerate(p.group_by) if j != idx]
        us = ts.data.to(torch.int64)
        if isinstance(ts.dtype, T.DateType):
            us = us * 86_400_000_000
        if others:
            pgid, _, _np = group_ids(others)
        else:
            pgid = torch.zeros(n, dtype=torch.int64, device=dev)
        order = torch.argsort(us)
        order = order[torch.argsort(pgid.index_select(0, order), stable=True)]
        g_s = pgid.index_select(0, order)
        t_s = us.index_select(0, order)
        new = torch.ones(n, dtype=torch.bool, device=dev)
        if n > 1:
            new[1:] = (g_s[1:] != g_s[:-1]) | ((t_s[1:] - t_s[:-1]) > gap)
        sid_sorted = torch.cumsum(new.to(torch.int64), 0) - 1
        ns = int(sid_sorted[-1].item()) + 1 if n else 0
        starts = torch.zeros(ns, dtype=torch.int64, device=dev)
        ends = torch.zeros(ns, dtype=torch.int64, device=dev)
        starts[sid_sorted[new]] = t_s[new]
        # last row of each session: next row starts a new session (or EOF)
        last = torch.ones(n, dtype=torch.bool, device=dev)
        if n > 1:
            last[:-1] = new[1:]
        ends[sid_sorted[last]] = t_s[last] + gap
        sid = torch.empty(n, dtype=torch.int64, device=dev)
        sid[order] = sid_sorted
        return StructColumn(
            [("start", Column(T.TIMESTAMP, starts.index_select(0, sid),
                              ts.validity)),
             ("end", Column(T.TIMESTAMP, ends.index_select(0, sid),
                            ts.validity))],
            ts.validity, dtype=e.dtype)

    def _grouping_sets_aggregate(self, p: S.Aggregate) -> Chunk:
        """ROLLUP/CUBE/GROUPING SETS: one aggregation per set, absent keys
        null, results unioned; grouping()/grouping_id() computed per set
        (ref: Spark semantics; sail-function grouping_id)."""
        child = self._gather(self.execute(p.input))
        n, dev = child.num_rows, child.device
        key_cols = [broadcast(self.ev.eval(g, child), n, dev) for g in p.group_by]
        nkeys = len(key_cols)
        parts: List[Chunk] = []
        from .eval import Scalar

        for gset in p.grouping_sets:
            included = sorted(set(gset))
            sub_keys = [key_cols[i] for i in included]
            if n == 0:
                continue
            if sub_keys:
                gid, rep, ng = group_ids(sub_keys)
            else:
                gid, ng = global_ids(n, dev)
                rep = torch.zeros(1, dtype=torch.int64, device=dev)
            out_keys = []
            for i in range(nkeys):
                if i in included:
                    out_keys.append(key_cols[i].gather(rep))
                else:
                    out_keys.append(broadcast(Scalar(None, p.group_by[i].dtype), ng, dev))
            gbits = 0
            for i in range(nkeys):
                if i not in included:
                    gbits |= 1 << (nkeys - 1 - i)
            agg_cols = []
            for a in p.aggs:
                if a.name in ("grouping", "grouping_id"):
                    if a.name == "grouping":
                        pos = a.args[0].value
                        v = 0 if pos in included else 1
                        agg_cols.append(Column(T.I32, torch.full((ng,), v, dtype=torch.int32, device=dev)))
                    else:
                        agg_cols.append(Column(T.I64, torch.full((ng,), gbits, dtype=torch.int64, device=dev)))
                    continue
                args = [broadcast(self.ev.eval(x, child), n, dev) for x in a.args]
                fmask = self.ev.eval_mask(a.filter, child) if a.filter is not None else None
                agg_cols.append(agg_eval(a.name, args, gid, ng, a.distinct, fmask, a.dtype))
            parts.append(Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema]))
        if not parts:
            return Chunk([_empty_agg_col(a, dev) for a in p.aggs], [nm for nm, _ in p.schema])
        cols = [concat_columns([pt.columns[i] for pt in parts])
                for i in range(len(p.schema))]
        return Chunk(cols, [nm for nm, _ in p.schema])

    def _try_masked_aggregate(self, p: S.Aggregate) -> Optional[Chunk]:
        from ..plan.rules.util import substitute_refs
        from .aggregates import MaskedGroupsUnsupported, fused_agg_batch

        node = p.input
        while isinstance(node, S.SubqueryAlias):  # identity wrappers
            node = node.input
        proj = None
        if isinstance(node, S.Project) and not any(
                isinstance(e, (S.ScalarSubquery, S.Exists, S.InSubquery))
                for pe in node.exprs for e in pe.walk()):
            f = node.input
            while isinstance(f, S.SubqueryAlias):
                f = f.input
            if not isinstance(f, S.Filter):
                return None
            proj = node
        elif isinstance(node, S.Filter):
            f = node
        else:
            return None
        if self.dctx is not None:
            return None  # distributed path handles its own exchange first

        base = self.execute(f.input)
        n = base.num_rows
        dev = base.device
        if n == 0:
            return None  # empty: use the standard path

        mask = self.ev.eval_mask(f.condition, base)

        def rebased(e: S.Expr) -> S.Expr:
            return substitute_refs(e, proj.exprs) if proj is not None else e

        # selectivity decides: high -> keep the mask (no materialization);
        # low -> gather only the referenced columns once and drop the mask
        n_pass = int(mask.sum().item())
        if n_pass == 0:
            return None

        def materialize():
            nonlocal base, n, mask
            from ..plan.rules.util import expr_refs

            refs = set()
            for g in p.group_by:
                refs |= expr_refs(rebased(g))
            for a in p.aggs:
                for x in a.args:
                    refs |= expr_refs(rebased(x))
                if a.filter is not None:
                    refs |= expr_refs(rebased(a.filter))
            idx = torch.nonzero(mask, as_tuple=False).squeeze(1)
            cols: List[Optional[Column]] = [None] * len(base.columns)
            for i in refs:
                cols[i] = base.columns[i].gather(idx)
            base = Chunk(cols, list(base.names), base.partitioning)
            base.forced_rows = n_pass
            n = n_pass
            mask = None

        # CPU reference path always materializes (masked per-agg indexing is
        # slower there); on GPU keep the mask unless selectivity is low
        if n_pass < 0.2 * n or not base.device.type == "cuda":
            materialize()

        key_cols = None
        if p.group_by:
            key_exprs = [rebased(g) for g in p.group_by]
            key_cols = [broadcast(self.ev.eval(g, base), n, dev) for g in key_exprs]
        try:
            if key_cols is not None:
                gid, rep, ng = group_ids(key_cols, mask=mask)
            else:
                gid, ng = global_ids(n, dev)
        except MaskedGroupsUnsupported:
            materialize()
            key_cols = [broadcast(self.ev.eval(g, base), n, dev)
                        for g in [rebased(g) for g in p.group_by]]
            gid, rep, ng = group_ids(key_cols, mask=None)
        out_keys = [c.gather(rep) for c in key_cols] if key_cols is not None else []

        args_list = []
        fmasks = []
        for a in p.aggs:
            args_list.append([broadcast(self.ev.eval(rebased(x), base), n, dev)
                              for x in a.args])
            if a.filter is not None:
                fm = self.ev.eval_mask(rebased(a.filter), base)
                if mask is not None:
                    fm = fm & mask
            else:
                fm = mask
            fmasks.append(fm)
        agg_cols = fused_agg_batch(p.aggs, args_list, fmasks, gid, ng)
        if agg_cols is None:
            agg_cols = [agg_eval(a.name, args, gid, ng, a.distinct, fmask, a.dtype)
                        for a, args, fmask in zip(p.aggs, args_list, fmasks)]
        return Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema])

    def _dist_aggregate(self, p: S.Aggregate, child: Chunk) -> Chunk:
        """Two-phase distributed aggregate: local partials -> all_gather the
        (tiny) partial tables -> merge. High-cardinality groups instead
        SHUFFLE the input by group key (each group lands whole on one rank,
        full local aggregation, output stays sharded) — gathered partials at
        1e8+ groups would replicate the whole table. Falls back to gathering
        the input for non-decomposable aggregates over low-cardinality
        groups (DISTINCT, percentiles, ...)."""
        from ..exec.distributed import decompose_agg, gather_chunk, shuffle_chunk

        if p.group_by:
            n0, dev0 = child.num_rows, child.device
            key_cols0 = [broadcast(self.ev.eval(g, child), n0, dev0) for g in p.group_by]
            if n0:
                _, _, ng_local = group_ids(key_cols0)
            else:
                ng_local = 0
            threshold = int(self.ctx.session.conf.get(
                "sail.exec.agg_shuffle_threshold_groups", str(4_000_000)))
            # CONSENSUS across ranks (shards differ; a rank-local count would
            # send ranks down different exchange paths — gloo/RCCL mismatch)
            ng_consensus = self.dctx.consensus_sum(
                ng_local * self.dctx.world) // self.dctx.world  # mean estimate
            if ng_consensus * self.dctx.world > threshold:
                ncols = len(child.columns)
                ext = Chunk(list(child.columns) + key_cols0,
                            list(child.names) + [f"__k{i}" for i in range(len(key_cols0))],
                            "sharded")
                shuffled = shuffle_chunk(ext, list(range(ncols, ncols + len(key_cols0))),
                                         self.dctx)
                base = Chunk(shuffled.columns[:ncols], list(child.names), "replicated")
                out = self._local_aggregate_with_keys(
                    p, base, shuffled.columns[ncols:])
                out.partitioning = "sharded"
                return out

        decomps = [decompose_agg(a) for a in p.aggs]
        if any(d is None for d in decomps):
            child = self._gather(child)
            p2 = S.Aggregate(input=None, group_by=p.group_by, aggs=p.aggs)
            p2.schema = p.schema
            return self._local_aggregate(p2, child)

        n, dev = child.num_rows, child.device
        # 1) local partials
        if p.group_by:
            key_cols = [broadcast(self.ev.eval(g, child), n, dev) for g in p.group_by]
            if n:
                gid, rep, ng = group_ids(key_cols)
                local_keys = [c.gather(rep) for c in key_cols]
            else:
                gid, ng = torch.zeros(0, dtype=torch.int64, device=dev), 0
                local_keys = key_cols
        else:
            if n:
                gid, ng = global_ids(n, dev)
            else:
                gid, ng = torch.zeros(0, dtype=torch.int64, device=dev), 0
            local_keys = []
        partial_cols = []
        partial_merges = []
        for a, d in zip(p.aggs, decomps):
            args = [broadcast(self.ev.eval(x, child), n, dev) for x in a.args]
            fmask = self.ev.eval_mask(a.filter, child) if a.filter is not None else None
            for pname, mname in zip(d.partials, d.merges):
                use_args = args if (args or pname != "count") else []
                col = agg_eval(pname, use_args, gid, ng, False, fmask,
                               None) if ng else _empty_partial(pname, dev)
                partial_cols.append(col)
                partial_merges.append(mname)
        # 2) exchange partial tables
        partial = Chunk(local_keys + partial_cols,
                        [f"k{i}" for i in range(len(local_keys))]
                        + [f"p{i}" for i in range(len(partial_cols))], "sharded")
        gathered = gather_chunk(partial, self.dctx)
        # 3) merge
        gn = gathered.num_rows
        nk = len(local_keys)
        if p.group_by:
            if gn == 0:
                return Chunk(gathered.columns[:nk]
                             + [_empty_agg_col(a, dev) for a in p.aggs],
                             [nm for nm, _ in p.schema])
            mgid, mrep, mng = group_ids(gathered.columns[:nk])
            out_keys = [c.gather(mrep) for c in gathered.columns[:nk]]
        else:
            mgid, mng = global_ids(gn, dev) if gn else (torch.zeros(0, dtype=torch.int64, device=dev), 1)
            out_keys = []
        merged = []
        for i, mname in enumerate(partial_merges):
            col = gathered.columns[nk + i]
            merged.append(agg_eval(mname, [col], mgid, mng, False, None, None))
        out_cols = []
        ci = 0
        for a, d in zip(p.aggs, decomps):
            k = len(d.partials)
            out_cols.append(d.finalize(merged[ci:ci + k], a.dtype))
            ci += k
        return Chunk(out_keys + out_cols, [nm for nm, _ in p.schema])

    def _local_aggregate_with_keys(self, p: S.Aggregate, child: Chunk,
                                   key_cols) -> Chunk:
        """Local aggregation with pre-evaluated group-key columns."""
        n, dev = child.num_rows, child.device
        if n == 0:
            return Chunk([c.slice(0, 0) for c in key_cols]
                         + [_empty_agg_col(a, dev) for a in p.aggs],
                         [nm for nm, _ in p.schema])
        args_list = [[broadcast(self.ev.eval(x, child), n, dev) for x in a.args]
                     for a in p.aggs]
        fmasks = [self.ev.eval_mask(a.filter, child) if a.filter is not None else None
                  for a in p.aggs]
        from .aggregates import fused_agg_batch, try_direct_aggregate

        d = try_direct_aggregate(key_cols, p.aggs, args_list, fmasks)
        if d is not None:
            out_keys, agg_cols = d
            return Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema])
        gid, rep, ng = group_ids(key_cols)
        out_keys = [c.gather(rep) for c in key_cols]
        agg_cols = fused_agg_batch(p.aggs, args_list, fmasks, gid, ng)
        if agg_cols is None:
            agg_cols = [agg_eval(a.name, args, gid, ng, a.distinct, fmask, a.dtype)
                        for a, args, fmask in zip(p.aggs, args_list, fmasks)]
        return Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema])

    def _local_aggregate(self, p: S.Aggregate, child: Chunk) -> Chunk:
        n = child.num_rows
        dev = child.device
        if p.group_by:
            key_cols = [broadcast(self.ev.eval(g, child), n, dev) for g in p.group_by]
            if n == 0:
                return Chunk(key_cols + [_empty_agg_col(a, dev) for a in p.aggs],
                             [nm for nm, _ in p.schema])
            gid, rep, ng = group_ids(key_cols)
            out_keys = [c.gather(rep) for c in key_cols]
        else:
            if n == 0:
                cols = [_empty_global_agg(a, dev) for a in p.aggs]
                return Chunk(cols, [nm for nm, _ in p.schema])
            gid, ng = global_ids(n, dev)
            out_keys = []
        args_list = [[broadcast(self.ev.eval(x, child), n, dev) for x in a.args]
                     for a in p.aggs]
        fmasks = [self.ev.eval_mask(a.filter, child) if a.filter is not None else None
                  for a in p.aggs]
        from .aggregates import fused_agg_batch

        agg_cols = fused_agg_batch(p.aggs, args_list, fmasks, gid, ng)
        if agg_cols is None:
            agg_cols = [agg_eval(a.name, args, gid, ng, a.distinct, fmask, a.dtype)
                        for a, args, fmask in zip(p.aggs, args_list, fmasks)]
        return Chunk(out_keys + agg_cols, [nm for nm, _ in p.schema])

    # -- joins -------------------------------------------------------------
    def _x_Join(self, p: S.Join) -> Chunk:
        left = self.execute(p.left)
        right = self.execute(p.right)
        if self.dctx is not None:
            lp, rp = left.partitioning, right.partitioning
            if lp == "sharded" or rp == "sharded":
                out_part = "sharded"
                if rp == "sharded" and lp == "sharded":
                    if self._should_shuffle(p, left, right):
                        left, right = self._shuffle_join_inputs(p, left, right)
                    else:
                        # broadcast join: replicate the (smaller) build side
                        right = self._gather(right)
                elif lp == "replicated" and rp == "sharded":
                    if p.how not in ("inner", "cross"):
                        # left-side semantics need every left row exactly once
                        right = self._gather(right)
                        out_part = "replicated"
                out = join_chunks(self.ev, p, left, right)
                out.partitioning = out_part
                return out
        return join_chunks(self.ev, p, left, right)

    def _should_shuffle(self, p: S.Join, left: Chunk, right: Chunk) -> bool:
        """Shuffle both sides instead of broadcasting the build side when the
        replicated build side would exceed the broadcast budget (xGMI has the
        bandwidth, HBM capacity is the constraint at SF1000 scale)."""
        if p.how not in ("inner", "semi", "anti", "left"):
            return False
        from .executor import split_join_condition

        equi, _ = split_join_condition(p.on, len(left.columns)) if p.on is not None else ([], None)
        if not equi:
            return False
        threshold = int(self.ctx.session.conf.get(
            "sail.exec.broadcast_threshold_bytes", str(2 << 30)))
        # CONSENSUS: shard sizes differ per rank, but every rank must pick
        # the same exchange plan (divergent paths = collective mismatch) —
        # decide on the global build-side size via one tiny all_reduce.
        return self.dctx.consensus_sum(_chunk_bytes(right)) > threshold

    def _shuffle_join_inputs(self, p: S.Join, left: Chunk, right: Chunk):
        from ..exec.distributed import shuffle_chunk
        from .executor import split_join_condition

        equi, _ = split_join_condition(p.on, len(left.columns))
        lkeys = [i for i, _ in equi]
        rkeys = [j - len(left.columns) for _, j in equi]
        return (shuffle_chunk(left, lkeys, self.dctx),
                shuffle_chunk(right, rkeys, self.dctx))

    # -- set ops -----------------------------------------------------------
    def _x_SetOp(self, p: S.SetOp) -> Chunk:
        left = self.execute(p.left)
        right = self.execute(p.right)
        # align right columns to left schema types
        rcols = []
        for i, (n_, t_) in enumerate(p.schema):
            c = right.columns[i]
            if c.dtype != t_:
                c = cast_column(c, t_)
            rcols.append(c)
        lcols = []
        for i, (n_, t_) in enumerate(p.schema):
            c = left.columns[i]
            if c.dtype != t_:
                c = cast_column(c, t_)
            lcols.append(c)
        names = [n_ for n_, _ in p.schema]
        if p.op == "union":
            cols = [concat_columns([a, b]) for a, b in zip(lcols, rcols)]
            out = Chunk(cols, names)
            if not p.is_all:
                gid, rep, ng = group_ids(out.columns)
                out = out.gather(rep)
            return out
        lchunk = Chunk(lcols, names)
        rchunk = Chunk(rcols, names)
        _, _, counts = equi_join(rchunk.columns, lchunk.columns, "semi")
        if p.op == "intersect":
            mask = counts > 0
        else:  # except
            mask = counts == 0
        out = lchunk.filter_mask(mask)
        if not p.is_all:
            gid, rep, ng = group_ids(out.columns)
            out = out.gather(rep)
        return out

    # -- window ------------------------------------------------------------
    def _x_WindowPlan(self, p: S.WindowPlan) -> Chunk:
        from .window import eval_window

        child = self.execute(p.input)
        if self.dctx is not None and child.partitioning == "sharded" \
                and p.window_exprs:
            wes = [we.child if isinstance(we, S.Alias) else we
                   for we in p.window_exprs]
            parts = [tuple(repr(x) for x in e.partition_by) for e in wes]
            if parts[0] and all(pt == parts[0] for pt in parts) \
                    and self.dctx.consensus_sum(child.num_rows) \
                    >= self.DIST_SORT_MIN_ROWS:
                # partitioned windows: shuffle so each
                # PARTITION BY group is wholly on one rank, evaluate
                # locally, stay sharded — no whole-table gather
                from ..exec.distributed import (partition_ids,
                                                shuffle_chunk_by_pids)

                n = child.num_rows
                key_cols = [broadcast(self.ev.eval(x, child), n, child.device)
                            for x in wes[0].partition_by]
                pids = partition_ids(key_cols, self.dctx.world)
                local = shuffle_chunk_by_pids(child, pids, self.dctx)
                cols = list(local.columns)
                for e in wes:
                    cols.append(eval_window(self.ev, e, local))
                return Chunk(cols, [nm for nm, _ in p.schema], "sharded")
        child = self._gather(child)
        cols = list(child.columns)
        for we in p.window_exprs:
            e = we.child if isinstance(we, S.Alias) else we
            cols.append(eval_window(self.ev, e, child))
        return Chunk(cols, [n for n, _ in p.schema])

    # -- commands ----------------------------------------------------------
    def _x_CreateView(self, p: S.CreateView) -> Chunk:
        self.ctx.session.catalog.create_view(p.name, p.input, replace=p.replace)
        return Chunk([], [])

    def _x_CreateTable(self, p: S.CreateTable) -> Chunk:
        sess = self.ctx.session
        head, _, rest = p.name.partition(".")
        path = None
        if head.lower() in ("parquet", "csv", "json", "delta", "iceberg") and rest:
            path, fmt = rest, head.lower()
        elif p.location and (p.format or "").lower() in ("parquet", "csv",
                                                         "json", "delta",
                                                         "iceberg"):
            path, fmt = p.location, p.format.lower()
        if path is not None:
            # CREATE TABLE <fmt>.`/path` [USING fmt] AS SELECT ... /
            # CREATE TABLE t USING fmt LOCATION '/path' AS SELECT ...
            if p.input is None:
                raise ExecError("path-based CREATE TABLE requires AS SELECT")
            from ..datasource.registry import write_source

            data = self.execute(p.input)
            write_source(fmt, path, data,
                         "overwrite" if p.replace else "error", p.options, None)
            if path != p.name and not p.name.count("."):
                # register the named table as a view over the path
                view = S.DataSourceRead(format=fmt, paths=[path])
                view.schema = [(n, c.dtype) for n, c in
                               zip(data.names, data.columns)]
                view.__dict__["_table_name"] = p.name
                sess.catalog.create_view(p.name, view, replace=True)
                prov = getattr(sess.catalog, "persistent", None)
                if prov is not None:
                    from ..catalogs.persistent import TableDef

                    prov.create_table(TableDef(
                        p.name, fmt, path, schema=view.schema,
                        options=dict(p.options or {})), replace=True)
            return Chunk([], [])
        if p.input is not None:
            data = self.execute(p.input)
            sess.catalog.register_table_chunk(p.name, data, [(n, t) for n, t in p.input.schema])
        else:
            sess.catalog.create_empty_table(p.name, p.columns)
        return Chunk([], [])

    def _x_DropTable(self, p: S.DropTable) -> Chunk:
        cat = self.ctx.session.catalog
        cat.drop(p.name, if_exists=p.if_exists)
        prov = getattr(cat, "persistent", None)
        if prov is not None:
            prov.drop_table(p.name.split(".")[-1], if_exists=True)
        return Chunk([], [])

    def _x_InsertInto(self, p: S.InsertInto) -> Chunk:
        data = self.execute(p.input)
        self.ctx.session.catalog.insert_into(p.table, data, overwrite=p.overwrite)
        return Chunk([], [])

    def _x_Write(self, p: S.Write) -> Chunk:
        from ..datasource.registry import write_source

        data = self.execute(p.input)
        write_source(p.format, p.path, data, p.mode, p.options, p.partition_by)
        return Chunk([], [])

    def _x_MergeInto(self, p) -> Chunk:
        from .dml import execute_merge

        return execute_merge(self, p)

    def _x_UpdateTable(self, p) -> Chunk:
        from .dml import execute_update

        return execute_update(self, p)

    def _x_DeleteFrom(self, p) -> Chunk:
        from .dml import execute_delete

        return execute_delete(self, p)

    def _x_Explain(self, p: S.Explain) -> Chunk:
        if p.mode == "analyze":
            # EXPLAIN ANALYZE: execute under the tracer, report per-operator
            # wall time / rows (TracingExec analogue — ref:
            # crates/sail-telemetry/src/execution/physical_plan.rs:54)
            from ..utils.trace import Tracer

            sub = ExecutionContext(self.ctx.session, self.ctx.device)
            sub.tracer = Tracer(self.ctx.device)
            out = Executor(sub).execute(p.input)
            text = (S.plan_tree_string(p.input).rstrip()
                    + "\n\n== Analyzed (wall times) ==\n"
                    + sub.tracer.trace.render()
                    + f"\n\nresult rows: {out.num_rows}")
            return Chunk([StringColumn.from_pylist([text])], ["plan"])
        text = S.plan_tree_string(p.input)
        return Chunk([StringColumn.from_pylist([text])], ["plan"])

    def _x_SetConfig(self, p: S.SetConfig) -> Chunk:
        if p.value is not None:
            self.ctx.session.conf[p.key] = p.value
        val = self.ctx.session.conf.get(p.key)
        return Chunk([StringColumn.from_pylist([p.key]), StringColumn.from_pylist([val])],
                     ["key", "value"])

    @staticmethod
    def _delta_path(name: str) -> str:
        head, _, rest = name.partition(".")
        if head.lower() != "delta" or not rest:
            raise ExecError(f"{name}: expected delta.`/path`")
        return rest

    def _x_VacuumTable(self, p: S.VacuumTable) -> Chunk:
        from ..datasource.delta import DeltaLog

        log = DeltaLog(self._delta_path(p.name))
        removed = log.vacuum(p.retention_hours if p.retention_hours is not None
                             else 168.0, p.dry_run)
        return Chunk([StringColumn.from_pylist(removed, dict_encode=False)],
                     ["removed_file"])

    def _x_DescribeHistory(self, p: S.DescribeHistory) -> Chunk:
        from ..datasource.delta import DeltaLog

        rows = DeltaLog(self._delta_path(p.name)).history()
        return Chunk([
            Column.from_values([r["version"] for r in rows], T.I64),
            Column.from_values([r["timestamp_ms"] for r in rows], T.I64),
            StringColumn.from_pylist([r["operation"] for r in rows], dict_encode=False),
            Column.from_values([r["num_added_files"] for r in rows], T.I64),
            Column.from_values([r["num_removed_files"] for r in rows], T.I64),
        ], ["version", "timestamp_ms", "operation", "num_added_files",
            "num_removed_files"])

    def _x_AlterTable(self, p: S.AlterTable) -> Chunk:
        cat = self.ctx.session.catalog
        key = cat._key(p.name)
        t = cat._tables.get(key)
        schema = list(cat._schemas.get(key) or [])
        if p.action == "add_columns":
            nrows = t.num_rows if t is not None else 0
            for cn, ct in p.columns:
                if any(n.lower() == cn.lower() for n, _ in schema):
                    raise ExecError(f"column {cn} already exists")
                schema.append((cn, ct))
                if t is not None:
                    t.columns[cn] = Column.from_values([None] * nrows, ct,
                                                       device="cpu")
        elif p.action == "drop_column":
            schema = [(n, ty) for n, ty in schema if n.lower() != p.column.lower()]
            if t is not None:
                for n in list(t.columns):
                    if n.lower() == p.column.lower():
                        del t.columns[n]
        elif p.action == "rename_column":
            schema = [(p.new_name if n.lower() == p.column.lower() else n, ty)
                      for n, ty in schema]
            if t is not None:
                t.columns = {(p.new_name if n.lower() == p.column.lower() else n): c
                             for n, c in t.columns.items()}
        elif p.action == "rename_table":
            nk = cat._key(p.new_name)
            if t is not None:
                cat._tables[nk] = cat._tables.pop(key)
            cat._schemas[nk] = cat._schemas.pop(key, schema)
            if key in cat._replicated:
                cat._replicated.discard(key)
                cat._replicated.add(nk)
            return Chunk([StringColumn.from_pylist(
                [f"renamed {p.name} to {p.new_name}"], dict_encode=False)], ["result"])
        cat._schemas[key] = schema
        cat._col_stats = {k: v for k, v in cat._col_stats.items() if k[0] != key}
        return Chunk([StringColumn.from_pylist([f"altered {p.name}"], dict_encode=False)],
                     ["result"])

    def _x_ShowFunctions(self, p: S.ShowFunctions) -> Chunk:
        from ..functions.registry import AGG_FUNCTIONS, SCALAR_RETURN, WINDOW_FUNCTIONS
        from .aggregates import UDAFS

        names = sorted(set(SCALAR_RETURN) | AGG_FUNCTIONS | WINDOW_FUNCTIONS
                       | set(self.ctx.session.udfs) | set(UDAFS))
        if p.pattern:
            import fnmatch

            pat = p.pattern.strip("'\"")
            names = [n for n in names if fnmatch.fnmatch(n, pat.replace("%", "*"))]
        return Chunk([StringColumn.from_pylist(names, dict_encode=False)], ["function"])

    def _x_ShowDatabases(self, p: S.ShowDatabases) -> Chunk:
        names = sorted(self.ctx.session.catalog._databases)
        return Chunk([StringColumn.from_pylist(names, dict_encode=False)],
                     ["namespace"])

    def _x_ShowCatalogs(self, p: S.ShowCatalogs) -> Chunk:
        return Chunk([StringColumn.from_pylist(["spark_catalog"],
                                               dict_encode=False)],
                     ["catalog"])

    def _x_ShowColumns(self, p: S.ShowColumns) -> Chunk:
        schema = self.ctx.session.catalog.table_schema(p.name)
        if schema is None:
            raise ValueError(f"table not found: {p.name}")
        return Chunk([StringColumn.from_pylist([n for n, _ in schema],
                                               dict_encode=False)],
                     ["col_name"])

    def _x_ShowCreateTable(self, p: S.ShowCreateTable) -> Chunk:
        cat = self.ctx.session.catalog
        schema = cat.table_schema(p.name)
        if schema is None:
            raise ValueError(f"table not found: {p.name}")
        cols = ",\n  ".join(f"{n} {T.type_name(t).upper()}"
                            for n, t in schema)
        stmt = f"CREATE TABLE {p.name} (\n  {cols}\n)"
        c = cat._comments.get(cat._key(p.name))
        if c:
            stmt += f"\nCOMMENT '{c}'"
        return Chunk([StringColumn.from_pylist([stmt], dict_encode=False)],
                     ["createtab_stmt"])

    def _x_ShowViews(self, p: S.ShowViews) -> Chunk:
        cat = self.ctx.session.catalog
        names = sorted(cat._views)
        if p.pattern:
            import fnmatch

            pat = p.pattern.strip("'\"").replace("%", "*")
            names = [n for n in names if fnmatch.fnmatch(n, pat)]
        return Chunk([
            StringColumn.from_pylist([cat.current_database] * len(names),
                                     dict_encode=False),
            StringColumn.from_pylist(names, dict_encode=False),
            Column(T.BOOL, torch.ones(len(names), dtype=torch.bool))],
            ["namespace", "viewName", "isTemporary"])

    def _x_ShowPartitions(self, p: S.ShowPartitions) -> Chunk:
        # engine tables are whole-partition columnar: no hive partitions
        return Chunk([StringColumn.from_pylist([], dict_encode=False)],
                     ["partition"])

    def _x_ShowTblProperties(self, p: S.ShowTblProperties) -> Chunk:
        cat = self.ctx.session.catalog
        props = dict(cat._tbl_properties.get(cat._key(p.name), {}))
        c = cat._comments.get(cat._key(p.name))
        if c:
            props.setdefault("comment", c)
        ks = sorted(props)
        return Chunk([
            StringColumn.from_pylist(ks, dict_encode=False),
            StringColumn.from_pylist([props[k] for k in ks],
                                     dict_encode=False)],
            ["key", "value"])

    def _x_UseDatabase(self, p: S.UseDatabase) -> Chunk:
        cat = self.ctx.session.catalog
        if p.name.lower() not in cat._databases:
            raise ValueError(f"database not found: {p.name}")
        cat.current_database = p.name.lower()
        return Chunk([], [])

    def _x_CreateDatabase(self, p: S.CreateDatabase) -> Chunk:
        cat = self.ctx.session.catalog
        nm = p.name.lower()
        if nm in cat._databases and not p.if_not_exists:
            raise ValueError(f"database already exists: {p.name}")
        cat._databases.add(nm)
        prov = getattr(cat, "persistent", None)
        if prov is not None:
            prov.create_database(nm, if_not_exists=True,
                                 comment=p.comment)
        return Chunk([], [])

    def _x_DropDatabase(self, p: S.DropDatabase) -> Chunk:
        cat = self.ctx.session.catalog
        nm = p.name.lower()
        if nm not in cat._databases:
            if p.if_exists:
                return Chunk([], [])
            raise ValueError(f"database not found: {p.name}")
        if nm == "default":
            raise ValueError("cannot drop the default database")
        cat._databases.discard(nm)
        if nm == cat.current_database:
            cat.current_database = "default"
        prov = getattr(cat, "persistent", None)
        if prov is not None:
            try:
                prov.drop_database(nm, cascade=p.cascade)
            except ValueError:
                if not p.if_exists:
                    raise
        return Chunk([], [])

    def _x_RefreshTable(self, p: S.RefreshTable) -> Chunk:
        # drop cached parquet page indexes + device dictionaries + stats
        from ..datasource import gpu_parquet as G

        G._INDEX_CACHE.clear()
        G._DICT_CACHE.clear()
        cat = self.ctx.session.catalog
        k = cat._key(p.name)
        cat._col_stats = {key: v for key, v in cat._col_stats.items()
                          if key[0] != k}
        return Chunk([StringColumn.from_pylist([f"refreshed {p.name}"],
                                               dict_encode=False)],
                     ["result"])

    def _x_TruncateTable(self, p: S.TruncateTable) -> Chunk:
        cat = self.ctx.session.catalog
        k = cat._key(p.name)
        schema = cat.table_schema(p.name)
        from .column import Column as _C

        cols = {}
        for n, t in schema:
            cols[n] = _C.from_values([], t, device="cpu")
        cat.register_table(p.name, Table(cols), list(schema))
        return Chunk([], [])

    def _x_CommentOn(self, p: S.CommentOn) -> Chunk:
        cat = self.ctx.session.catalog
        k = cat._key(p.name)
        if p.comment is None:
            cat._comments.pop(k, None)
        else:
            cat._comments[k] = p.comment
        return Chunk([], [])

    def _x_DescribeQuery(self, p: S.DescribeQuery) -> Chunk:
        rows = [(n, T.type_name(t), "") for n, t in p.input.schema]
        return Chunk([
            StringColumn.from_pylist([r[0] for r in rows],
                                     dict_encode=False),
            StringColumn.from_pylist([r[1] for r in rows],
                                     dict_encode=False),
            StringColumn.from_pylist([r[2] for r in rows],
                                     dict_encode=False)],
            ["col_name", "data_type", "comment"])

    def _x_CacheTable(self, p: S.CacheTable) -> Chunk:
        chunk = self.execute(p.input)
        cat = self.ctx.session.catalog
        # a same-name view would shadow the materialized table: stash it so
        # UNCACHE can restore the logical definition
        key = cat._key(p.name)
        vp = cat._views.pop(key, None)
        if vp is not None:
            cat.__dict__.setdefault("_cached_views", {})[key] = vp
        cat.register_table(p.name, chunk.to_table(),
                           [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)])
        return Chunk([StringColumn.from_pylist([f"cached {p.name}"], dict_encode=False)],
                     ["result"])

    def _x_UncacheTable(self, p: S.UncacheTable) -> Chunk:
        cat = self.ctx.session.catalog
        key = cat._key(p.name)
        cat._tables.pop(key, None)
        cat._schemas.pop(key, None)
        stash = cat.__dict__.get("_cached_views", {})
        if key in stash:
            cat._views[key] = stash.pop(key)
        return Chunk([StringColumn.from_pylist([f"uncached {p.name}"], dict_encode=False)],
                     ["result"])

    def _x_AnalyzeTable(self, p: S.AnalyzeTable) -> Chunk:
        cat = self.ctx.session.catalog
        sch = cat.table_schema(p.name) or []
        cols = p.columns if p.columns else [n for n, _ in sch]
        if p.columns == []:
            cols = [n for n, _ in sch]
        names, rows_l, ndv_l = [], [], []
        for cname in cols:
            st = cat.column_stats(p.name, cname)
            names.append(cname)
            rows_l.append(st[0] if st else -1)
            ndv_l.append(st[1] if st and st[1] is not None else -1)
        return Chunk([StringColumn.from_pylist(names, dict_encode=False),
                      Column.from_values(rows_l, T.I64),
                      Column.from_values(ndv_l, T.I64)],
                     ["column", "rows", "ndv"])

    def _x_ShowTables(self, p: S.ShowTables) -> Chunk:
        names = self.ctx.session.catalog.list_tables()
        return Chunk([StringColumn.from_pylist([""] * len(names), dict_encode=False),
                      StringColumn.from_pylist(names, dict_encode=False),
                      Column.from_values([True] * len(names), T.BOOL)],
                     ["namespace", "tableName", "isTemporary"])

    def _x_DescribeTable(self, p: S.DescribeTable) -> Chunk:
        schema = self.ctx.session.catalog.table_schema(p.name)
        if schema is None:
            raise ExecError(f"table not found: {p.name}")
        return Chunk([StringColumn.from_pylist([n for n, _ in schema], dict_encode=False),
                      StringColumn.from_pylist([repr(t) for _, t in schema], dict_encode=False),
                      StringColumn.from_pylist([""] * len(schema), dict_encode=False)],
                     ["col_name", "data_type", "comment"])


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------

def sort_indices(ev: Evaluator, keys: List[S.SortKey], chunk: Chunk) -> torch.Tensor:
    """Stable multi-key argsort: iterate keys last-to-first with stable sorts.
    Null ordering per Spark: NULLS FIRST for ASC, NULLS LAST for DESC
    unless overridden."""
    n = chunk.num_rows
    dev = chunk.device
    idx = torch.arange(n, device=dev)
    for k in reversed(keys):
        col = broadcast(ev.eval(k.child, chunk), n, dev)
        keyvals = _sortable(col)
        keyvals = keyvals[idx]
        nulls_first = k.nulls_first if k.nulls_first is not None else k.ascending
        if col.validity is not None:
            vm = col.valid_mask()[idx]
            big = _null_sentinel(keyvals, nulls_first == k.ascending)
            keyvals = torch.where(vm, keyvals, big)
        order = torch.argsort(keyvals, stable=True, descending=not k.ascending)
        idx = idx[order]
    return idx


def _sortable(col: Column) -> torch.Tensor:
    if isinstance(col, StringColumn):
        from .joins import normalize_key

        if col.is_dict:
            return col.codes.to(torch.int64)  # dict is sorted => codes ordered
        # raw strings: host DENSE rank (equal values must get equal ranks or
        # stable multi-key sorting breaks on the later keys)
        vals = col.to_pylist()
        order = sorted(range(len(vals)), key=lambda i: (vals[i] is None, vals[i] or ""))
        rank = [0] * len(vals)
        r = -1
        prev = object()
        for i in order:
            if vals[i] != prev:
                r += 1
                prev = vals[i]
            rank[i] = r
        return torch.tensor(rank, dtype=torch.int64, device=col.device)
    d = col.data
    if d.dtype == torch.bool:
        return d.to(torch.int8)
    return d


def _null_sentinel(vals: torch.Tensor, lo: bool):
    if vals.dtype.is_floating_point:
        v = float("-inf") if lo else float("inf")
    else:
        info = torch.iinfo(vals.dtype)
        v = info.min if lo else info.max
    return torch.full_like(vals, v)


def concat_columns(cols: List[Column]) -> Column:
    c0 = cols[0]
    from .column import ListColumn

    if isinstance(c0, ListColumn):
        children = concat_columns([c.child for c in cols])
        offs = [cols[0].offsets]
        base = cols[0].offsets[-1]
        for c in cols[1:]:
            offs.append(c.offsets[1:] + base)
            base = base + c.offsets[-1]
        validity = None
        if any(c.validity is not None for c in cols):
            validity = torch.cat([c.valid_mask() for c in cols]).to(torch.uint8)
        return ListColumn(torch.cat(offs), children, validity, c0.dtype)
    if isinstance(c0, StringColumn):
        return _concat_strings(cols)
    from .column import StructColumn

    if isinstance(c0, StructColumn):
        names = [nm for nm, _ in c0.children_]
        kids = [(nm, concat_columns([dict(c.children_)[nm] for c in cols]))
                for nm in names]
        validity = None
        if any(c.validity is not None for c in cols):
            validity = torch.cat([c.valid_mask() for c in cols]).to(torch.uint8)
        return StructColumn(kids, validity, dtype=c0.dtype)
    data = torch.cat([c.data for c in cols])
    if any(c.validity is not None for c in cols):
        validity = torch.cat([c.valid_mask() for c in cols]).to(torch.uint8)
    else:
        validity = None
    return Column(c0.dtype, data, validity)


def _concat_strings(cols: List[StringColumn]) -> StringColumn:
    """Device-native string concat. All-dictionary inputs merge through a
    sorted union dictionary (host work touches only the small dictionaries;
    codes remap on device — keeps the engine's sorted-dict invariant);
    otherwise raw offsets/bytes are chained on device. Replaces a
    to_pylist/from_pylist host round-trip that dominated MERGE rewrites."""
    dev = cols[0].device
    total_dict = sum(c.offsets.shape[0] - 1 for c in cols if c.codes is not None)
    if all(c.codes is not None for c in cols) and total_dict <= 200_000:
        vals_list = [c.dict_values() for c in cols]
        union = sorted(set().union(*vals_list))
        pos = {s: i for i, s in enumerate(union)}
        from .column import _pack_strings

        offs, byts = _pack_strings(union, dev)
        new_codes = []
        for c, vals in zip(cols, vals_list):
            m = torch.tensor([pos[v] for v in vals] or [0], dtype=torch.int32,
                             device=dev)
            cc = c.codes
            remapped = m[cc.clamp_min(0).to(torch.int64)]
            new_codes.append(torch.where(cc >= 0, remapped,
                                         torch.full_like(cc, -1)))
        validity = None
        if any(c.validity is not None for c in cols):
            validity = torch.cat([c.valid_mask() for c in cols]).to(torch.uint8)
        return StringColumn(offs, byts, validity, torch.cat(new_codes),
                            dtype=cols[0].dtype)
    raws = [c.decode_dict() for c in cols]
    offs_parts = []
    bytes_parts = []
    shift = 0
    for r in raws:
        offs_parts.append(r.offsets[:-1] + shift)
        bytes_parts.append(r.bytes_)
        shift += int(r.offsets[-1].item())
    offs_parts.append(torch.tensor([shift], dtype=torch.int64, device=dev))
    validity = None
    if any(c.validity is not None for c in cols):
        validity = torch.cat([c.valid_mask() for c in cols]).to(torch.uint8)
    return StringColumn(torch.cat(offs_parts), torch.cat(bytes_parts),
                        validity, None, dtype=cols[0].dtype)


def join_chunks(ev: Evaluator, p: S.Join, left: Chunk, right: Chunk) -> Chunk:
    """Execute a join given both sides. Equi-conditions go through the
    hash/sort join; residual non-equi predicates are applied to the matched
    pairs (and fixed up for outer joins)."""
    how = p.how
    names = [n for n, _ in p.schema]
    if how == "cross" or p.on is None:
        if how not in ("cross", "inner"):
            raise ExecError(f"{how} join requires a condition")
        nl, nr = left.num_rows, right.num_rows
        li = torch.arange(nl, device=left.device).repeat_interleave(nr)
        ri = torch.arange(nr, device=left.device).repeat(nl)
        lcols = [c.gather(li) for c in left.columns]
        rcols = [c.gather(ri) for c in right.columns]
        return Chunk(lcols + rcols, names)

    equi, residual = split_join_condition(p.on, len(left.columns))

    if not equi:
        return _nested_loop_join(ev, p, left, right, names)

    lkeys = [left.columns[i] for i, _ in equi]
    rkeys = [right.columns[j - len(left.columns)] for _, j in equi]

    # orient: build on right, probe on left
    probe_idx, build_idx, counts = equi_join(rkeys, lkeys, how)

    if residual is not None and probe_idx.shape[0] > 0:
        # gather only the columns the residual references (a q21-style
        # self-join can have ~B matched pairs; full-width gathers OOM)
        from ..plan.rules.util import expr_refs, remap_expr

        nl = len(left.columns)
        refs = sorted(expr_refs(residual))
        remap = {old: new for new, old in enumerate(refs)}
        pair_cols = []
        for old in refs:
            if old < nl:
                pair_cols.append(left.columns[old].gather(probe_idx))
            else:
                pair_cols.append(right.columns[old - nl].gather(build_idx))
        pairs = Chunk(pair_cols, [f"c{i}" for i in range(len(pair_cols))])
        rmask = ev.eval_mask(remap_expr(residual, remap), pairs)
        del pairs, pair_cols
        probe_idx = probe_idx[rmask]
        build_idx = build_idx[rmask]
        if how in ("semi", "anti", "left", "full"):
            counts = torch.zeros(left.num_rows, dtype=torch.int64, device=left.device)
            counts.index_add_(0, probe_idx, torch.ones(probe_idx.shape[0], dtype=torch.int64, device=left.device))

    dev = left.device
    if how == "inner":
        return Chunk([c.gather(probe_idx) for c in left.columns]
                     + [c.gather(build_idx) for c in right.columns], names)
    if how == "semi":
        mask = counts > 0
        return left.filter_mask(mask)
    if how == "anti":
        mask = counts == 0
        return left.filter_mask(mask)
    if how in ("left", "full", "right"):
        unmatched_l = torch.nonzero(counts == 0, as_tuple=False).squeeze(1)
        li = torch.cat([probe_idx, unmatched_l])
        ri = torch.cat([build_idx, torch.full((unmatched_l.shape[0],), -1, dtype=torch.int64, device=dev)])
        if how == "right" or how == "full":
            matched_r = torch.zeros(right.num_rows, dtype=torch.bool, device=dev)
            if build_idx.shape[0]:
                matched_r[build_idx] = True
            unmatched_r = torch.nonzero(~matched_r, as_tuple=False).squeeze(1)
            if how == "right":
                li = torch.cat([probe_idx, torch.full((unmatched_r.shape[0],), -1, dtype=torch.int64, device=dev)])
                ri = torch.cat([build_idx, unmatched_r])
            else:
                li = torch.cat([li, torch.full((unmatched_r.shape[0],), -1, dtype=torch.int64, device=dev)])
                ri = torch.cat([ri, unmatched_r])
        lcols = [_gather_nullable(c, li) for c in left.columns]
        rcols = [_gather_nullable(c, ri) for c in right.columns]
        return Chunk(lcols + rcols, names)
    if how in ("rightsemi", "rightanti"):
        matched_r = torch.zeros(right.num_rows, dtype=torch.bool, device=dev)
        if build_idx.shape[0]:
            matched_r[build_idx] = True
        mask = matched_r if how == "rightsemi" else ~matched_r
        return right.filter_mask(mask)
    raise ExecError(f"join type {how}")


def _gather_nullable(c: Column, idx: torch.Tensor) -> Column:
    """Gather where idx == -1 produces NULL."""
    if len(c) == 0:
        # all-null fill from an EMPTY side (outer join against an empty
        # table): there is no row 0 to clamp to
        return Column.from_values([None] * int(idx.shape[0]), c.dtype,
                                  device=idx.device)
    null = idx < 0
    safe = torch.where(null, torch.zeros_like(idx), idx)
    out = c.gather(safe)
    if bool(null.any()):
        v = out.valid_mask() & ~null
        if isinstance(out, StringColumn):
            out.validity = v.to(torch.uint8)
            return out
        return Column(out.dtype, out.data, v.to(torch.uint8))
    return out


def _nested_loop_join(ev: Evaluator, p: S.Join, left: Chunk, right: Chunk, names):
    nl, nr = left.num_rows, right.num_rows
    dev = left.device
    li = torch.arange(nl, device=dev).repeat_interleave(nr)
    ri = torch.arange(nr, device=dev).repeat(nl)
    pairs = Chunk([c.gather(li) for c in left.columns] + [c.gather(ri) for c in right.columns],
                  [f"c{i}" for i in range(len(left.columns) + len(right.columns))])
    mask = ev.eval_mask(p.on, pairs)
    how = p.how
    if how == "inner":
        return Chunk([c for c in pairs.filter_mask(mask).columns], names)
    counts = torch.zeros(nl, dtype=torch.int64, device=dev)
    counts.index_add_(0, li[mask], torch.ones(int(mask.sum()), dtype=torch.int64, device=dev))
    if how == "semi":
        return left.filter_mask(counts > 0)
    if how == "anti":
        return left.filter_mask(counts == 0)
    rcounts = torch.zeros(nr, dtype=torch.int64, device=dev)
    rcounts.index_add_(0, ri[mask],
                       torch.ones(int(mask.sum()), dtype=torch.int64, device=dev))
    if how == "rightsemi":
        return right.filter_mask(rcounts > 0)
    if how == "rightanti":
        return right.filter_mask(rcounts == 0)
    if how in ("left", "right", "full"):
        keep_li = li[mask]
        keep_ri = ri[mask]
        lidx, ridx = keep_li, keep_ri
        if how in ("left", "full"):
            unmatched = torch.nonzero(counts == 0, as_tuple=False).squeeze(1)
            lidx = torch.cat([lidx, unmatched])
            ridx = torch.cat([ridx, torch.full((unmatched.shape[0],), -1,
                                               dtype=torch.int64, device=dev)])
        if how in ("right", "full"):
            runmatched = torch.nonzero(rcounts == 0, as_tuple=False).squeeze(1)
            lidx = torch.cat([lidx, torch.full((runmatched.shape[0],), -1,
                                               dtype=torch.int64, device=dev)])
            ridx = torch.cat([ridx, runmatched])
        lcols = [_gather_nullable(c, lidx) for c in left.columns]
        rcols = [_gather_nullable(c, ridx) for c in right.columns]
        return Chunk(lcols + rcols, names)
    raise ExecError(f"nested loop join type {how} TODO")


def split_join_condition(on: S.Expr, n_left: int):
    """Split an ON condition into equi-key pairs [(left_idx, right_idx)] and a
    residual expression (or None). A conjunct qualifies as an equi-key when it
    is BoundRef == BoundRef with sides on opposite inputs."""
    conjuncts = []

    def flatten(e):
        if isinstance(e, S.BinaryOp) and e.op == "and":
            flatten(e.left)
            flatten(e.right)
        else:
            conjuncts.append(e)

    flatten(on)
    equi = []
    residual = []
    for c in conjuncts:
        if isinstance(c, S.BinaryOp) and c.op == "=":
            l, r = c.left, c.right
            # strip widening int casts (int32->int64): normalize_key makes the
            # raw columns join-compatible; decimal rescales must stay.
            l = l.child if (isinstance(l, S.Cast) and isinstance(l.child, S.BoundRef)
                            and l.child.dtype is not None and l.child.dtype.is_integer
                            and l.dtype is not None and l.dtype.is_integer) else l
            r = r.child if (isinstance(r, S.Cast) and isinstance(r.child, S.BoundRef)
                            and r.child.dtype is not None and r.child.dtype.is_integer
                            and r.dtype is not None and r.dtype.is_integer) else r
            if isinstance(l, S.BoundRef) and isinstance(r, S.BoundRef):
                if l.index < n_left <= r.index:
                    equi.append((l.index, r.index))
                    continue
                if r.index < n_left <= l.index:
                    equi.append((r.index, l.index))
                    continue
        residual.append(c)
    res = None
    for c in residual:
        res = c if res is None else S.BinaryOp("and", res, c, T.BOOL)
    return equi, res


def _empty_agg_col(a: S.AggFunc, dev) -> Column:
    if isinstance(a.dtype, T.StringType):
        return StringColumn.from_pylist([], device=dev)
    return Column(a.dtype, torch.zeros(0, dtype=a.dtype.storage or torch.int64, device=dev))


def _empty_global_agg(a: S.AggFunc, dev) -> Column:
    if a.name in ("count", "count_if"):
        return Column(T.I64, torch.zeros(1, dtype=torch.int64, device=dev))
    if isinstance(a.dtype, T.StringType):
        return StringColumn.from_pylist([None], device=dev)
    return Column(a.dtype, torch.zeros(1, dtype=a.dtype.storage or torch.int64, device=dev),
                  torch.zeros(1, dtype=torch.uint8, device=dev))


def _empty_partial(pname: str, dev) -> Column:
    if pname in ("count", "count_if", "sum"):
        return Column(T.I64, torch.zeros(0, dtype=torch.int64, device=dev))
    return Column(T.F64, torch.zeros(0, dtype=torch.float64, device=dev))


def _chunk_bytes(chunk: Chunk) -> int:
    total = 0
    for c in chunk.columns:
        if c is None:
            continue
        if isinstance(c, StringColumn):
            total += int(c.bytes_.numel()) + 8 * len(c)
        else:
            total += c.data.numel() * c.data.element_size()
    return total
