"""Distributed operator strategies (SPMD, one process per GPU).

Every rank holds a shard of each big table (datagen/tpch._shard) and runs
the same plan; this module supplies the exchange points:

  * gather_chunk — all_gather a sharded chunk -> replicated chunk
    (broadcast joins, global sorts/limits)
  * partial/final aggregate decomposition — local partial aggregates are
    all_gathered (tiny) and merged, replacing the reference's shuffle-based
    two-phase aggregation (ref: crates/sail-execution InputMode::Shuffle /
    Merge) with collectives over xGMI.

Exchange payloads are whole columns (few large buffers), matching xGMI's
preference for large point-to-point transfers over many small messages.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..engine import types as T
from ..engine.chunk import Chunk
from ..engine.column import Column, StringColumn
from ..plan import spec as S
from .context import DistContext


def gather_column(c: Column, d: DistContext) -> Column:
    """NOTE: the validity mask is exchanged UNCONDITIONALLY (ones when the
    shard has no nulls): whether a shard contains nulls is rank-local state,
    and skipping the collective on some ranks only is a collective-order
    mismatch (caught by the world=4 gloo test). n uint8 per column is noise
    next to the data payload."""
    from ..engine.executor import concat_columns

    if isinstance(c, StringColumn):
        raw = c.decode_dict()
        offs_list = d.all_gather_tensors(raw.offsets[1:] - raw.offsets[:-1])
        bytes_list = d.all_gather_tensors(raw.bytes_)
        val_list = d.all_gather_tensors(raw.valid_mask().to(torch.uint8))
        cols = []
        for i in range(d.world):
            lens = offs_list[i]
            offs = torch.zeros(lens.shape[0] + 1, dtype=torch.int64, device=lens.device)
            torch.cumsum(lens, 0, out=offs[1:])
            v = val_list[i] if not bool(val_list[i].all()) else None
            cols.append(StringColumn(offs, bytes_list[i], v, None, dtype=c.dtype))
        return concat_columns(cols)
    data_list = d.all_gather_tensors(c.data)
    val_list = d.all_gather_tensors(c.valid_mask().to(torch.uint8))
    validity = torch.cat(val_list)
    if bool(validity.all()):
        validity = None
    return Column(c.dtype, torch.cat(data_list), validity)


def gather_chunk(chunk: Chunk, d: DistContext) -> Chunk:
    return Chunk([gather_column(c, d) for c in chunk.columns], list(chunk.names))


# ---------------------------------------------------------------------------
# aggregate decomposition
# ---------------------------------------------------------------------------

class AggDecomposition:
    """Partial aggregates + a merge recipe for one logical aggregate.

    partials: list of (agg_name, use_original_args: bool) executed locally;
    `finalize(cols)` combines the gathered partial columns (one Column per
    partial, already re-aggregated by group with the merge agg) into the
    final output column.
    """

    def __init__(self, partials, merges, finalize):
        self.partials = partials  # names for local partial aggs
        self.merges = merges      # agg names used to re-aggregate partials
        self.finalize = finalize  # fn(list[Column], out_type) -> Column


def decompose_agg(a: S.AggFunc) -> Optional[AggDecomposition]:
    name = a.name
    if a.distinct:
        return None  # handled by input gather fallback
    if name in ("sum", "try_sum"):
        return AggDecomposition(["sum"], ["sum"], lambda cols, t: _retype(cols[0], t))
    if name in ("count", "count_if"):
        return AggDecomposition([name], ["sum"], lambda cols, t: _retype(cols[0], T.I64))
    if name in ("min", "max", "any", "bool_and"):
        return AggDecomposition([name], [name], lambda cols, t: _retype(cols[0], t))
    if name == "avg":
        def fin(cols, t):
            from ..engine.aggregates import _avg_result

            sums, cnt = cols[0], cols[1]
            return _avg_result(sums.data, cnt.data, sums.dtype, t)

        return AggDecomposition(["sum", "count"], ["sum", "sum"], fin)
    if name in ("stddev_samp", "stddev_pop", "var_samp", "var_pop"):
        def fin(cols, t, _name=name):
            cnt = cols[0].data.to(torch.float64)
            s1 = cols[1].data.to(torch.float64)
            s2 = cols[2].data.to(torch.float64)
            mean = s1 / cnt.clamp_min(1)
            m2 = (s2 - cnt * mean * mean).clamp_min(0)
            denom = cnt - (1.0 if _name.endswith("_samp") else 0.0)
            var = m2 / denom.clamp_min(1e-300)
            data = torch.sqrt(var) if _name.startswith("stddev") else var
            valid = denom > 0
            return Column(T.F64, data,
                          None if bool(valid.all()) else valid.to(torch.uint8))

        return AggDecomposition(["count", "sumf", "sumsq"], ["sum", "sum", "sum"], fin)
    if name in ("first", "any_value"):
        return AggDecomposition(["first"], ["first"], lambda cols, t: cols[0])
    if name == "last":
        return AggDecomposition(["last"], ["last"], lambda cols, t: cols[0])
    return None


def _retype(c: Column, t) -> Column:
    if t is None or c.dtype == t:
        return c
    if isinstance(c, StringColumn):
        return c
    data = c.data
    if t.storage is not None and data.dtype != t.storage:
        data = data.to(t.storage)
    return Column(t, data, c.validity)


# ---------------------------------------------------------------------------
# hash-shuffle exchange (RCCL all_to_all over xGMI)
# ---------------------------------------------------------------------------

def _mix64(x: torch.Tensor) -> torch.Tensor:
    x = x ^ (x >> 33)
    x = x * -49064778989728563  # 0xFF51AFD7ED558CCD as signed int64
    x = x ^ (x >> 33)
    return x


def partition_ids(key_cols, world: int) -> torch.Tensor:
    """Hash-partition assignment for a set of join-key columns. Must be
    computed identically on every rank and both join sides: keys are packed
    with the cross-side-stable normalizer (ints/dates/decimals raw; raw
    strings FNV; dict strings decoded hash)."""
    from ..engine.joins import fnv_key_tensor, normalize_key
    from ..engine.column import StringColumn

    acc = None
    for c in key_cols:
        if isinstance(c, StringColumn):
            k = fnv_key_tensor(c.decode_dict())
        else:
            k = normalize_key(c)
        acc = k if acc is None else _mix64(acc * 31 + k)
    return torch.remainder(_mix64(acc), world)


def _exchange_1d(t: torch.Tensor, send_counts: torch.Tensor, d: DistContext) -> torch.Tensor:
    """all_to_all a 1-D tensor already grouped by destination rank.
    gloo (CPU tests) has no all_to_all: emulated with all_gather."""
    send_counts_l = [int(x) for x in send_counts.tolist()]
    if d.dist.get_backend() == "nccl":
        recv_counts = torch.zeros(d.world, dtype=torch.int64, device=t.device)
        sc = send_counts.to(t.device)
        d.dist.all_to_all_single(recv_counts, sc)
        recv_l = [int(x) for x in recv_counts.tolist()]
        out = torch.empty(sum(recv_l), dtype=t.dtype, device=t.device)
        d.dist.all_to_all_single(out, t, recv_l, send_counts_l)
        return out
    # gloo emulation: gather all (send tensor + counts), pick my slice
    parts = d.all_gather_tensors(t)
    counts = d.all_gather_tensors(send_counts.to(t.device if t.is_cuda else "cpu"))
    outs = []
    for src in range(d.world):
        cl = [int(x) for x in counts[src].tolist()]
        start = sum(cl[: d.rank])
        outs.append(parts[src][start : start + cl[d.rank]])
    return torch.cat(outs)


def shuffle_chunk(chunk: Chunk, key_idx, d: DistContext) -> Chunk:
    """Repartition a sharded chunk by hash of the key columns: one
    all_to_all per column payload (few large messages for xGMI's
    point-to-point links, vs the reference's many Flight streams)."""
    pids = partition_ids([chunk.columns[i] for i in key_idx], d.world)
    return shuffle_chunk_by_pids(chunk, pids, d)


def shuffle_chunk_by_pids(chunk: Chunk, pids: torch.Tensor, d: DistContext) -> Chunk:
    """Repartition a sharded chunk by an explicit per-row destination-rank
    tensor (hash partitioning for joins/aggregates, RANGE partitioning for
    the distributed sort). The stable argsort keeps same-destination rows in
    their rank-local order so ties stay deterministic."""
    world = d.world
    order = torch.argsort(pids, stable=True)
    send_counts = torch.bincount(pids, minlength=world).to(torch.int64)
    out_cols = []
    for c in chunk.columns:
        if isinstance(c, StringColumn):
            raw = c.decode_dict()
            raw = raw.gather(order)
            lens = raw.offsets[1:] - raw.offsets[:-1]
            new_lens = _exchange_1d(lens, send_counts, d)
            # per-destination byte counts for the payload exchange
            bc = torch.zeros(world, dtype=torch.int64, device=lens.device)
            bc.index_add_(0, pids[order], lens)
            new_bytes = _exchange_1d(raw.bytes_, bc, d)
            offs = torch.zeros(new_lens.shape[0] + 1, dtype=torch.int64, device=new_lens.device)
            torch.cumsum(new_lens, 0, out=offs[1:])
            out_cols.append(StringColumn(offs, new_bytes,
                                         _exchange_validity(c, order, send_counts, d)))
        else:
            data = c.data.index_select(0, order)
            new_data = _exchange_1d(data, send_counts, d)
            out_cols.append(Column(c.dtype, new_data,
                                   _exchange_validity(c, order, send_counts, d)))
    out = Chunk(out_cols, list(chunk.names), "sharded")
    return out


def range_partition_ids(keys: torch.Tensor, d: DistContext,
                        descending: bool = False) -> torch.Tensor:
    """Destination ranks for the range-partitioned distributed sort:
    stride-sampled splitters (all_gathered, so identical on every rank);
    equal keys always map to one rank, which keeps multi-key ties local.
    ref: the reference sorts per-partition then SortPreservingMerge
    (sail-physical-optimizer EnforceSorting); here ranks own disjoint key
    ranges so a rank-order concat IS the global order."""
    n = keys.numel()
    step = max(n // 4096, 1)
    sample = keys[::step].contiguous()
    allsamp = torch.cat(d.all_gather_tensors(sample))
    allsamp, _ = torch.sort(allsamp)
    m = allsamp.numel()
    if m == 0 or d.world <= 1:
        return torch.zeros(n, dtype=torch.int64, device=keys.device)
    idxs = torch.tensor([min((m * r) // d.world, m - 1)
                         for r in range(1, d.world)], device=allsamp.device)
    splitters = allsamp.index_select(0, idxs).contiguous()
    pids = torch.searchsorted(splitters, keys)
    if descending:
        pids = (d.world - 1) - pids
    return pids


def _exchange_validity(c, order, send_counts, d):
    """Unconditional (see gather_column): null-presence is rank-local and
    must not gate a collective."""
    v = c.valid_mask().to(torch.uint8).index_select(0, order)
    out = _exchange_1d(v, send_counts, d)
    return None if bool(out.all()) else out


def sync_table_stats(session):
    """Make planner column statistics rank-identical: all_reduce the
    min/max/dict-size of every sharded table's columns ONCE at registration
    (planning itself stays communication-free, so a rank can never hang on
    stats). Must be called in the same order on every rank."""
    d = getattr(session, "dist", None)
    if d is None or d.world <= 1:
        return
    cat = session.catalog
    for name in sorted(cat._tables):
        if cat.is_replicated(name):
            continue  # identical on all ranks already
        t = cat._tables[name]
        grows = cat._global_rows.get(name, None)
        if grows is None:
            grows = d.consensus_sum(t.num_rows)
            cat._global_rows[name] = grows
        for cn, c in t.columns.items():
            rows = len(c)
            from ..engine.column import StringColumn

            if isinstance(c, StringColumn):
                local = c.dict_size if c.is_dict else -1
                m = d.consensus_max(local)
                ndv = m if m >= 0 else None
            elif rows == 0 or c.data.dtype == torch.bool:
                ndv = 2
            elif c.data.dtype.is_floating_point:
                ndv = None
            else:
                lo, hi = d.consensus_minmax(
                    int(c.data.min().item()) if rows else 0,
                    int(c.data.max().item()) if rows else 0)
                ndv = min(grows, hi - lo + 1)
            if ndv is not None:
                ndv = min(max(1, ndv), grows)
            cat.set_column_stats(name, cn, grows, ndv)
