"""Window function evaluation (torch path).

Partition/order by sort, then per-function segment ops
(ref: crates/sail-function/src/window/, DataFusion BoundedWindowAggExec role).
Supported: row_number, rank, dense_rank, percent_rank, cume_dist, ntile,
lag/lead, sum/count/min/max/avg over the default or running frame.
"""
from __future__ import annotations

from typing import List

import torch

from ..plan import spec as S
from . import types as T
from .chunk import Chunk
from .column import Column, StringColumn
from .eval import Evaluator, broadcast
from .joins import normalize_key


def eval_window(ev: Evaluator, e: S.WindowExpr, chunk: Chunk) -> Column:
    n = chunk.num_rows
    dev = chunk.device
    if n == 0:
        return Column(e.dtype, torch.zeros(0, dtype=(e.dtype.storage or torch.int64), device=dev))

    # 1) sort by (partition, order)
    from .executor import sort_indices

    keys = [S.SortKey(p, True, None) for p in e.partition_by] + list(e.order_by)
    idx = sort_indices(ev, keys, chunk) if keys else torch.arange(n, device=dev)
    sorted_chunk = chunk.gather(idx)

    # 2) partition boundaries in sorted order
    if e.partition_by:
        pcols = [broadcast(ev.eval(p, sorted_chunk), n, dev) for p in e.partition_by]
        pk = [normalize_key(c) for c in pcols]
        change = torch.zeros(n, dtype=torch.bool, device=dev)
        for k in pk:
            change[1:] |= k[1:] != k[:-1]
        change[0] = True
    else:
        change = torch.zeros(n, dtype=torch.bool, device=dev)
        change[0] = True
    part_id = torch.cumsum(change.to(torch.int64), 0) - 1
    part_start_pos = torch.nonzero(change, as_tuple=False).squeeze(1)
    pos_in_part = torch.arange(n, device=dev) - part_start_pos[part_id]

    # order-key ties (for rank)
    if e.order_by:
        ocols = [broadcast(ev.eval(k.child, sorted_chunk), n, dev) for k in e.order_by]
        ok = [normalize_key(c) for c in ocols]
        newval = change.clone()
        for k in ok:
            newval[1:] |= k[1:] != k[:-1]
        newval[0] = True
    else:
        newval = change

    f = e.func
    fname = f.name.lower() if isinstance(f, (S.Func, S.AggFunc)) else None
    out = None

    if fname == "row_number":
        out = Column(T.I32, (pos_in_part + 1).to(torch.int32))
    elif fname in ("rank", "dense_rank"):
        if fname == "rank":
            rank_at = torch.where(newval, pos_in_part + 1, torch.zeros_like(pos_in_part))
            run = torch.cummax(torch.where(newval, pos_in_part + 1, torch.zeros_like(pos_in_part))
                               + part_id * (2 * n), 0).values - part_id * (2 * n)
            out = Column(T.I32, run.to(torch.int32))
        else:
            dr = torch.cumsum(newval.to(torch.int64), 0)
            base = dr[part_start_pos[part_id]]
            out = Column(T.I32, (dr - base + 1).to(torch.int32))
    elif fname == "percent_rank":
        part_sizes = torch.zeros(int(part_id.max().item()) + 1, dtype=torch.int64, device=dev)
        part_sizes.index_add_(0, part_id, torch.ones(n, dtype=torch.int64, device=dev))
        run = torch.cummax(torch.where(newval, pos_in_part + 1, torch.zeros_like(pos_in_part))
                           + part_id * (2 * n), 0).values - part_id * (2 * n)
        denom = (part_sizes[part_id] - 1).clamp_min(1).to(torch.float64)
        out = Column(T.F64, (run - 1).to(torch.float64) / denom)
    elif fname in ("lag", "lead"):
        src = broadcast(ev.eval(f.args[0], sorted_chunk), n, dev)
        off = int(f.args[1].value) if len(f.args) > 1 and isinstance(f.args[1], S.Literal) else 1
        if fname == "lead":
            off = -off
        tgt = torch.arange(n, device=dev) - off
        valid = (tgt >= 0) & (tgt < n)
        tgt_part = torch.where(valid, part_id[tgt.clamp(0, n - 1)], torch.full_like(tgt, -1))
        valid &= tgt_part == part_id
        safe = tgt.clamp(0, n - 1)
        g = src.gather(safe)
        vmask = g.valid_mask() & valid
        default = None
        if len(f.args) > 2:
            d = f.args[2]
            if isinstance(d, S.UnaryOp) and d.op == "neg" \
                    and isinstance(d.child, S.Literal):
                default = -d.child.value
            elif isinstance(d, S.Literal) and d.value is not None:
                default = d.value
        if isinstance(g, StringColumn):
            if default is not None:
                # default fills only out-of-frame rows (Spark); in-frame
                # NULL values stay NULL
                vals = g.to_pylist()
                in_frame = valid.cpu().tolist()
                vals = [v if m else default
                        for v, m in zip(vals, in_frame)]
                out = StringColumn.from_pylist(vals, device=str(dev),
                                               dict_encode=False)
                out.validity = (g.valid_mask() | ~valid).to(torch.uint8)
            else:
                g.validity = vmask.to(torch.uint8)
                out = g
        elif default is not None:
            fill = torch.zeros_like(g.data)
            if g.data.is_floating_point():
                fill = fill + float(default)
            else:
                fill = fill + int(default)
            data = torch.where(valid, g.data, fill)
            validity = (g.valid_mask() | ~valid)
            out = Column(src.dtype, data,
                         None if bool(validity.all())
                         else validity.to(torch.uint8))
        else:
            out = Column(src.dtype, g.data, vmask.to(torch.uint8))
    elif fname == "cume_dist":
        part_sizes = torch.zeros(int(part_id.max().item()) + 1, dtype=torch.int64, device=dev)
        part_sizes.index_add_(0, part_id, torch.ones(n, dtype=torch.int64, device=dev))
        # number of rows <= current row's peer group (last peer position)
        peer_id = torch.cumsum(newval.to(torch.int64), 0) - 1
        npeers = int(peer_id.max().item()) + 1
        last_of_peer = torch.zeros(npeers, dtype=torch.int64, device=dev)
        last_of_peer.scatter_reduce_(0, peer_id, torch.arange(n, device=dev),
                                     reduce="amax", include_self=False)
        upto = last_of_peer[peer_id] - part_start_pos[part_id] + 1
        out = Column(T.F64, upto.to(torch.float64)
                     / part_sizes[part_id].to(torch.float64))
    elif fname == "nth_value":
        src = broadcast(ev.eval(f.args[0], sorted_chunk), n, dev)
        k = int(f.args[1].value)
        tgt = part_start_pos[part_id] + (k - 1)
        valid = pos_in_part >= (k - 1)
        g = src.gather(tgt.clamp(0, n - 1))
        vmask = g.valid_mask() & valid
        if isinstance(g, StringColumn):
            g.validity = vmask.to(torch.uint8)
            out = g
        else:
            out = Column(src.dtype, g.data, vmask.to(torch.uint8))
    elif fname == "ntile":
        buckets = int(f.args[0].value)
        part_sizes = torch.zeros(int(part_id.max().item()) + 1, dtype=torch.int64, device=dev)
        part_sizes.index_add_(0, part_id, torch.ones(n, dtype=torch.int64, device=dev))
        sz = part_sizes[part_id]
        out = Column(T.I32, (pos_in_part * buckets // sz + 1).to(torch.int32))
    elif isinstance(f, S.AggFunc):
        out = _window_agg(ev, f, sorted_chunk, part_id, pos_in_part, e, n, dev)
    if out is None:
        raise NotImplementedError(f"window function {fname}")

    # 3) scatter back to original row order
    inv = torch.empty_like(idx)
    inv[idx] = torch.arange(n, device=dev)
    return out.gather(inv)


def _window_agg(ev, f: S.AggFunc, sorted_chunk, part_id, pos_in_part, e, n, dev):
    """sum/count/avg/min/max/first/last over whole-partition, running, or
    bounded-ROWS frames (prefix-sum sliding windows)."""
    from .aggregates import agg_eval

    running = bool(e.order_by) and (e.frame is None or e.frame[1][0] == "unbounded_preceding")
    whole = not e.order_by or (e.frame is not None and e.frame[1][0] == "unbounded_preceding"
                               and e.frame[2][0] == "unbounded_following")
    bounded = (e.frame is not None and e.frame[0] == "rows"
               and not whole
               and e.frame[1][0] in ("preceding", "following", "current", "unbounded_preceding")
               and e.frame[2][0] in ("preceding", "following", "current"))
    ng = int(part_id.max().item()) + 1
    args = [broadcast(ev.eval(a, sorted_chunk), n, dev) for a in f.args] if f.args else []
    part_start_pos = torch.nonzero(
        torch.cat([torch.ones(1, dtype=torch.bool, device=dev),
                   part_id[1:] != part_id[:-1]]), as_tuple=False).squeeze(1)
    if bounded and f.name in ("sum", "count", "avg", "first", "first_value",
                              "last", "last_value"):
        return _bounded_rows_agg(f, args, part_id, pos_in_part,
                                 part_start_pos, e.frame, n, dev)
    range_bounded = (e.frame is not None and e.frame[0] == "range"
                     and not whole and len(e.order_by) == 1
                     and e.frame[1][0] in ("preceding", "following",
                                           "current", "unbounded_preceding")
                     and e.frame[2][0] in ("preceding", "following",
                                           "current",
                                           "unbounded_following"))
    if range_bounded and f.name in ("sum", "count", "avg", "first",
                                    "first_value", "last", "last_value"):
        ok = broadcast(ev.eval(e.order_by[0].child, sorted_chunk), n, dev)
        if ok.data is not None and not isinstance(ok, StringColumn):
            if not e.order_by[0].ascending:
                raise NotImplementedError(
                    "RANGE frame over a descending key")
            return _bounded_range_agg(f, args, part_id, part_start_pos,
                                      e.frame, ok.data, n, dev)
    if f.name in ("first", "first_value") and running:
        src = args[0]
        g = src.gather(part_start_pos[part_id])
        return g
    if f.name in ("last", "last_value") and running:
        return args[0]  # Spark: running frame's last_value IS the current row
    if whole:
        per_group = agg_eval(f.name, args, part_id, ng, f.distinct, None, f.dtype)
        return per_group.gather(part_id)
    if running:
        # cumulative within partition
        if f.name == "count":
            return Column(T.I64, pos_in_part + 1)
        c = args[0]
        x = c.data.to(torch.float64 if c.dtype.is_float else torch.int64)
        cum = torch.cumsum(x, 0)
        part_start = torch.nonzero(torch.cat([torch.ones(1, dtype=torch.bool, device=dev),
                                              part_id[1:] != part_id[:-1]]), as_tuple=False).squeeze(1)
        base = torch.where(part_start[part_id] > 0, cum[(part_start[part_id] - 1).clamp(0)],
                           torch.zeros_like(cum[0]).expand(n) if cum.dim() else torch.zeros(n, dtype=cum.dtype, device=dev))
        run = cum - base
        if f.name == "sum":
            return Column(f.dtype, run if not isinstance(f.dtype, T.DecimalType) else run.to(torch.int64))
        if f.name == "avg":
            return Column(T.F64, run.to(torch.float64) / (pos_in_part + 1).to(torch.float64))
        if f.name in ("min", "max"):
            opped = torch.cummin if f.name == "min" else torch.cummax
            # segment cumulative min/max: reset at partition starts via offset trick
            big = x.max() - x.min() + 1 if n else 1
            biased = x + part_id * (big if f.name == "max" else -big)
            res = opped(biased, 0).values - part_id * (big if f.name == "max" else -big)
            return Column(f.dtype, res.to(args[0].data.dtype))
    raise NotImplementedError(f"window frame for {f.name}")


def _bounded_rows_agg(f, args, part_id, pos_in_part, part_start_pos, frame, n, dev):
    """ROWS BETWEEN a AND b sliding window via prefix sums: win = cum[t] -
    cum[s-1] with s/t clipped to the partition (one pass, no per-row loop)."""
    sizes = torch.zeros(int(part_id.max().item()) + 1, dtype=torch.int64, device=dev)
    sizes.index_add_(0, part_id, torch.ones(n, dtype=torch.int64, device=dev))
    pstart = part_start_pos[part_id]
    pend = pstart + sizes[part_id] - 1
    i = torch.arange(n, device=dev)

    def bound_pos(b, default_lo):
        kind, v = b
        if kind == "unbounded_preceding":
            return pstart
        if kind == "current":
            return i
        if kind == "preceding":
            return i - int(v)
        if kind == "following":
            return i + int(v)
        return pstart if default_lo else pend

    s_pos = torch.maximum(bound_pos(frame[1], True), pstart)
    t_pos = torch.minimum(bound_pos(frame[2], False), pend)
    return _frame_agg(f, args, s_pos, t_pos, pstart, pend, n, dev)


def _bounded_range_agg(f, args, part_id, part_start_pos, frame, order_vals,
                       n, dev):
    """RANGE BETWEEN a AND b: value-based bounds over the (single, numeric)
    ORDER BY key. The input is sorted by (partition, key), so a
    partition-biased searchsorted finds each row's frame as an index
    range; the prefix-sum machinery then applies unchanged."""
    sizes = torch.zeros(int(part_id.max().item()) + 1, dtype=torch.int64,
                        device=dev)
    sizes.index_add_(0, part_id, torch.ones(n, dtype=torch.int64,
                                            device=dev))
    pstart = part_start_pos[part_id]
    pend = pstart + sizes[part_id] - 1
    ov = order_vals.to(torch.float64)
    span = float((ov.max() - ov.min()).item()) if n else 0.0

    def off(b):
        kind, v = b
        return float(v) if v is not None else 0.0

    bias_step = span + abs(off(frame[1])) + abs(off(frame[2])) + 1.0
    biased = ov + part_id.to(torch.float64) * bias_step

    def lo_pos(b):
        kind, v = b
        if kind == "unbounded_preceding":
            return pstart
        if kind == "current":
            return torch.searchsorted(biased, biased, side="left")
        d = float(v)
        tgt = biased - d if kind == "preceding" else biased + d
        return torch.searchsorted(biased, tgt, side="left")

    def hi_pos(b):
        kind, v = b
        if kind == "unbounded_following":
            return pend
        if kind == "current":
            return torch.searchsorted(biased, biased, side="right") - 1
        d = float(v)
        tgt = biased + d if kind == "following" else biased - d
        return torch.searchsorted(biased, tgt, side="right") - 1

    s_pos = torch.maximum(lo_pos(frame[1]), pstart)
    t_pos = torch.minimum(hi_pos(frame[2]), pend)
    return _frame_agg(f, args, s_pos, t_pos, pstart, pend, n, dev)


def _frame_agg(f, args, s_pos, t_pos, pstart, pend, n, dev):
    empty = s_pos > t_pos
    s_pos = torch.minimum(s_pos, pend)
    t_pos = torch.maximum(t_pos, pstart)

    if f.name in ("first", "first_value"):
        src = args[0]
        g = src.gather(s_pos)
        if empty.any():
            vm = g.valid_mask() & ~empty
            if isinstance(g, StringColumn):
                g.validity = vm.to(torch.uint8)
                return g
            return Column(src.dtype, g.data, vm.to(torch.uint8))
        return g
    if f.name in ("last", "last_value"):
        src = args[0]
        g = src.gather(t_pos)
        if empty.any():
            vm = g.valid_mask() & ~empty
            if isinstance(g, StringColumn):
                g.validity = vm.to(torch.uint8)
                return g
            return Column(src.dtype, g.data, vm.to(torch.uint8))
        return g

    c = args[0] if args else None
    valid = c.valid_mask() if c is not None else torch.ones(n, dtype=torch.bool, device=dev)
    vcnt = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(valid.to(torch.int64), 0, out=vcnt[1:])
    wcnt = (vcnt[t_pos + 1] - vcnt[s_pos]).clamp_min(0)
    wcnt = torch.where(empty, torch.zeros_like(wcnt), wcnt)
    if f.name == "count":
        return Column(T.I64, wcnt)
    x = c.data
    x = x.to(torch.float64) if c.dtype.is_float else x.to(torch.int64)
    x = torch.where(valid, x, torch.zeros_like(x))
    cum = torch.zeros(n + 1, dtype=x.dtype, device=dev)
    torch.cumsum(x, 0, out=cum[1:])
    wsum = cum[t_pos + 1] - cum[s_pos]
    vmask = (wcnt > 0)
    vv = None if bool(vmask.all()) else vmask.to(torch.uint8)
    if f.name == "avg":
        from .aggregates import _avg_result

        res = _avg_result(torch.where(vmask, wsum, torch.zeros_like(wsum)), wcnt,
                          c.dtype, f.dtype)
        if vv is not None:
            res = Column(res.dtype, res.data, vv)
        return res
    out_t = f.dtype or c.dtype
    data = wsum
    if isinstance(c.dtype, T.DecimalType):
        data = data.to(torch.int64)
    return Column(out_t, data, vv)
