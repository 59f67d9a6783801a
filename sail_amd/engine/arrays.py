"""Array (list) functions as segment ops over flat child columns.

Spark's array function family (ref: crates/sail-plan/src/function/scalar/
collection.rs names/semantics) implemented over ListColumn's
offsets+child layout: per-row work becomes one vectorized pass over the
flat child with `segment_ids()` as the reduction key — the same
whole-partition shape every other operator in this engine uses, so these
run unchanged on device. String-element ordering ops (sort_array/distinct
over strings) and regex split run on host (documented deviation).
"""
from __future__ import annotations

from typing import List

import torch

from . import types as T
from .column import Column, ListColumn, StringColumn

_BIG = (1 << 62)


def _bcast(v, chunk):
    from .eval import broadcast

    return broadcast(v, chunk.num_rows, chunk.device)


def _scalar_value(v):
    from .eval import Scalar

    return v.value if isinstance(v, Scalar) else None


def _f_array(args, out, chunk, ev):
    n, dev = chunk.num_rows, chunk.device
    elem_t = out.element
    from .eval import cast_column

    cols = [cast_column(_bcast(a, chunk), elem_t) for a in args]
    k = len(cols)
    if k == 0:
        offs = torch.zeros(n + 1, dtype=torch.int64, device=dev)
        return ListColumn(offs, Column.from_values([], elem_t, device=dev))
    from .executor import concat_columns

    allc = concat_columns(cols)  # column j occupies [j*n, (j+1)*n)
    p = torch.arange(n * k, dtype=torch.int64, device=dev)
    idx = (p % k) * n + (p // k)
    child = allc.gather(idx)
    offs = torch.arange(0, (n + 1) * k, k, dtype=torch.int64, device=dev)
    return ListColumn(offs, child)


def _f_size(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    data = (c.offsets[1:] - c.offsets[:-1]).to(torch.int32)
    return Column(T.I32, data, c.validity)


def _element_at(c: ListColumn, pos: torch.Tensor, one_based: bool):
    """pos per row; 1-based supports negative-from-end (Spark element_at)."""
    lens = c.lengths()
    if one_based:
        idx0 = torch.where(pos < 0, lens + pos, pos - 1)
    else:
        idx0 = pos
    ok = (idx0 >= 0) & (idx0 < lens) & c.valid_mask()
    safe = torch.where(ok, idx0, torch.zeros_like(idx0)) + c.offsets[:-1]
    safe = safe.clamp(0, max(len(c.child) - 1, 0))
    if len(c.child) == 0:
        out = Column.from_values([None] * len(c), c.child.dtype, device=c.device)
        return out
    got = c.child.gather(safe)
    valid = ok & got.valid_mask()
    v = None if bool(valid.all()) else valid.to(torch.uint8)
    if type(got) is not Column:  # String/Struct/List/Map keep their layout
        got.validity = v
        return got
    return Column(got.dtype, got.data, v)


def _f_element_at(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    iv = _scalar_value(args[1])
    if iv is not None:
        pos = torch.full((len(c),), int(iv), dtype=torch.int64, device=c.device)
    else:
        pos = _bcast(args[1], chunk).data.to(torch.int64)
    return _element_at(c, pos, one_based=True)


def _f_get(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    iv = _scalar_value(args[1])
    if iv is not None:
        pos = torch.full((len(c),), int(iv), dtype=torch.int64, device=c.device)
    else:
        pos = _bcast(args[1], chunk).data.to(torch.int64)
    return _element_at(c, pos, one_based=False)


def _elem_match(c: ListColumn, needle) -> torch.Tensor:
    """Per-child-row bool: element == needle (nulls never match)."""
    child = c.child
    if isinstance(child, StringColumn):
        from .joins import fnv_key_tensor

        key = fnv_key_tensor(child.decode_dict())
        nk = fnv_key_tensor(StringColumn.from_pylist([needle], device=c.device,
                                                     dict_encode=False))
        m = key == nk[0]
    else:
        from .eval import Scalar, cast_value

        sc = cast_value(Scalar(needle, child.dtype), child.dtype, None)
        m = child.data == _scalar_tensor(sc.value, child, c.device)
    return m & child.valid_mask()


def _scalar_tensor(v, child: Column, dev):
    if isinstance(child.dtype, T.DecimalType):
        v = int(round(float(v) * 10 ** child.dtype.scale))
    return torch.tensor(v, dtype=child.data.dtype, device=dev)


def _f_array_contains(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    needle = _scalar_value(args[1])
    m = _elem_match(c, needle).to(torch.int64)
    seg = c.segment_ids()
    acc = torch.zeros(len(c), dtype=torch.int64, device=c.device)
    acc.index_add_(0, seg, m)
    return Column(T.BOOL, acc > 0, c.validity)


def _f_array_position(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    needle = _scalar_value(args[1])
    m = _elem_match(c, needle)
    seg = c.segment_ids()
    pos_in_row = torch.arange(len(c.child), dtype=torch.int64, device=c.device) \
        - torch.repeat_interleave(c.offsets[:-1], c.lengths())
    cand = torch.where(m, pos_in_row + 1, torch.full_like(pos_in_row, _BIG))
    first = torch.full((len(c),), _BIG, dtype=torch.int64, device=c.device)
    first.scatter_reduce_(0, seg, cand, reduce="amin", include_self=True)
    return Column(T.I64, torch.where(first == _BIG, torch.zeros_like(first), first),
                  c.validity)


def _seg_minmax(c: ListColumn, is_min: bool):
    child = c.child
    if isinstance(child, (StringColumn, ListColumn)):
        raise NotImplementedError("array_min/max over non-primitive elements")
    seg = c.segment_ids()
    data = child.data
    if data.dtype == torch.bool:
        data = data.to(torch.int64)
    if data.dtype.is_floating_point:
        ext = torch.finfo(data.dtype).max
    else:
        ext = torch.iinfo(data.dtype).max
    fill = data.new_full((), ext if is_min else -ext)
    vals = torch.where(child.valid_mask(), data, fill.expand_as(data))
    acc = fill.expand(len(c)).clone()
    acc.scatter_reduce_(0, seg, vals, reduce="amin" if is_min else "amax",
                        include_self=True)
    cnt = torch.zeros(len(c), dtype=torch.int64, device=c.device)
    cnt.index_add_(0, seg, child.valid_mask().to(torch.int64))
    valid = (cnt > 0) & c.valid_mask()
    v = None if bool(valid.all()) else valid.to(torch.uint8)
    out_data = acc.to(child.data.dtype) if child.data.dtype != data.dtype else acc
    return Column(child.dtype, out_data, v)


def _f_array_min(args, out, chunk, ev):
    return _seg_minmax(_bcast(args[0], chunk), True)


def _f_array_max(args, out, chunk, ev):
    return _seg_minmax(_bcast(args[0], chunk), False)


def _f_sort_array(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    asc = True
    if len(args) > 1:
        asc = bool(_scalar_value(args[1]))
    child = c.child
    if isinstance(child, (StringColumn, ListColumn)):
        # host fallback for non-primitive elements
        vals = c.to_pylist()
        return ListColumn.from_pylist(
            [sorted(v, reverse=not asc) if v is not None else None for v in vals],
            child.dtype, device=c.device)
    seg = c.segment_ids()
    # stable two-pass: order by value, then stable by segment
    key = child.data.to(torch.float64) if child.data.dtype == torch.bool else child.data
    o1 = torch.argsort(key, stable=True, descending=not asc)
    o2 = torch.argsort(seg.index_select(0, o1), stable=True)
    order = o1.index_select(0, o2)
    return ListColumn(c.offsets, child.gather(order), c.validity, c.dtype)


def _f_array_distinct(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    child = c.child
    seg = c.segment_ids()
    from .joins import normalize_key

    key = normalize_key(child)
    # first-occurrence order: sort by (seg, key, pos); drop adjacent dups;
    # restore original position order among the kept
    n = len(child)
    pos = torch.arange(n, dtype=torch.int64, device=c.device)
    o1 = torch.argsort(key, stable=True)
    o2 = torch.argsort(seg.index_select(0, o1), stable=True)
    order = o1.index_select(0, o2)  # sorted by (seg, key), pos-stable
    sk = key.index_select(0, order)
    ss = seg.index_select(0, order)
    keep = torch.ones(n, dtype=torch.bool, device=c.device)
    if n > 1:
        keep[1:] = (sk[1:] != sk[:-1]) | (ss[1:] != ss[:-1])
    kept = order[keep]
    kept = kept.sort().values  # original order
    new_seg = seg.index_select(0, kept)
    lens = torch.bincount(new_seg, minlength=len(c))
    offs = torch.zeros(len(c) + 1, dtype=torch.int64, device=c.device)
    torch.cumsum(lens, 0, out=offs[1:])
    return ListColumn(offs, child.gather(kept), c.validity, c.dtype)


def _f_array_join(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    delim = _scalar_value(args[1])
    vals = c.to_pylist()
    joined = [None if v is None else delim.join(str(x) for x in v if x is not None)
              for v in vals]
    return StringColumn.from_pylist(joined, device=chunk.device)


def _f_slice(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    lens = c.lengths()
    start = _bcast(args[1], chunk).data.to(torch.int64)
    length = _bcast(args[2], chunk).data.to(torch.int64)
    s0 = torch.where(start > 0, start - 1, lens + start).clamp_min(0)
    take = torch.minimum(length.clamp_min(0), (lens - s0).clamp_min(0))
    new_off = torch.zeros(len(c) + 1, dtype=torch.int64, device=c.device)
    torch.cumsum(take, 0, out=new_off[1:])
    total = int(new_off[-1].item())
    pos = torch.arange(total, dtype=torch.int64, device=c.device) \
        - torch.repeat_interleave(new_off[:-1], take)
    child_idx = torch.repeat_interleave(c.offsets[:-1] + s0, take) + pos
    return ListColumn(new_off, c.child.gather(child_idx), c.validity, c.dtype)


def _f_sequence(args, out, chunk, ev):
    a = _bcast(args[0], chunk).data.to(torch.int64)
    b = _bcast(args[1], chunk).data.to(torch.int64)
    if len(args) > 2:
        st = _bcast(args[2], chunk).data.to(torch.int64)
    else:
        st = torch.where(b >= a, torch.ones_like(a), -torch.ones_like(a))
    lens = ((b - a) // st + 1).clamp_min(0)
    n = a.shape[0]
    offs = torch.zeros(n + 1, dtype=torch.int64, device=a.device)
    torch.cumsum(lens, 0, out=offs[1:])
    total = int(offs[-1].item())
    pos = torch.arange(total, dtype=torch.int64, device=a.device) \
        - torch.repeat_interleave(offs[:-1], lens)
    child = torch.repeat_interleave(a, lens) + pos * torch.repeat_interleave(st, lens)
    return ListColumn(offs, Column(T.I64, child))


def _f_split(args, out, chunk, ev):
    import re as _re

    c = _bcast(args[0], chunk)
    pat = _scalar_value(args[1])
    rx = _re.compile(pat)
    vals = c.to_pylist()
    parts = [None if v is None else rx.split(v) for v in vals]
    flat: List[str] = []
    lens = []
    for p in parts:
        if p is None:
            lens.append(0)
        else:
            lens.append(len(p))
            flat.extend(p)
    dev = chunk.device
    offs = torch.zeros(len(parts) + 1, dtype=torch.int64, device=dev)
    if parts:
        torch.cumsum(torch.tensor(lens, dtype=torch.int64, device=dev), 0, out=offs[1:])
    validity = None
    if any(p is None for p in parts):
        validity = torch.tensor([0 if p is None else 1 for p in parts],
                                dtype=torch.uint8, device=dev)
    child = StringColumn.from_pylist(flat, device=dev)
    return ListColumn(offs, child, validity)


def _f_arrays_overlap(args, out, chunk, ev):
    a = _bcast(args[0], chunk)
    b = _bcast(args[1], chunk)
    from .joins import fnv_key_tensor, normalize_key

    def _xkey(ch):
        if isinstance(ch, StringColumn):
            return fnv_key_tensor(ch.decode_dict())
        return normalize_key(ch)

    ka, kb = _xkey(a.child), _xkey(b.child)
    sa, sb = a.segment_ids(), b.segment_ids()
    # per row: any key of a present in b's keys for the same row — pack
    # (row, key) and intersect via sorted search
    pa = sa * 0x9E3779B97F4A7C15 + ka
    pb = sb * 0x9E3779B97F4A7C15 + kb
    sorted_b = torch.sort(pb).values
    idx = torch.searchsorted(sorted_b, pa)
    idx = idx.clamp_max(max(sorted_b.shape[0] - 1, 0))
    hit = (sorted_b.shape[0] > 0) & (sorted_b.index_select(0, idx) == pa) \
        & a.child.valid_mask()
    acc = torch.zeros(len(a), dtype=torch.int64, device=a.device)
    acc.index_add_(0, sa, hit.to(torch.int64))
    valid = a.valid_mask() & b.valid_mask()
    return Column(T.BOOL, acc > 0,
                  None if bool(valid.all()) else valid.to(torch.uint8))


IMPLS = {
    "array": _f_array, "size": _f_size, "cardinality": _f_size,
    "element_at": _f_element_at, "get": _f_get, "try_element_at": _f_element_at,
    "array_contains": _f_array_contains, "array_position": _f_array_position,
    "array_min": _f_array_min, "array_max": _f_array_max,
    "sort_array": _f_sort_array, "array_sort": _f_sort_array,
    "array_distinct": _f_array_distinct, "array_join": _f_array_join,
    "slice": _f_slice, "sequence": _f_sequence, "split": _f_split,
    "arrays_overlap": _f_arrays_overlap,
}


# ---------------------------------------------------------------------------
# map functions (MapColumn: offsets + parallel keys/values children)
# ---------------------------------------------------------------------------

def _f_map(args, out, chunk, ev):
    """map(k1, v1, k2, v2, ...)"""
    from .column import MapColumn
    from .eval import cast_column

    n, dev = chunk.num_rows, chunk.device
    kt, vt = out.key, out.value
    ks = [cast_column(_bcast(args[i], chunk), kt) for i in range(0, len(args), 2)]
    vs = [cast_column(_bcast(args[i], chunk), vt) for i in range(1, len(args), 2)]
    m = len(ks)
    from .executor import concat_columns

    if m == 0:
        offs = torch.zeros(n + 1, dtype=torch.int64, device=dev)
        return MapColumn(offs, Column.from_values([], kt, device=dev),
                         Column.from_values([], vt, device=dev))
    p = torch.arange(n * m, dtype=torch.int64, device=dev)
    idx = (p % m) * n + (p // m)
    keys = concat_columns(ks).gather(idx)
    values = concat_columns(vs).gather(idx)
    offs = torch.arange(0, (n + 1) * m, m, dtype=torch.int64, device=dev)
    return MapColumn(offs, keys, values)


def _f_map_from_arrays(args, out, chunk, ev):
    from .column import MapColumn

    ka = _bcast(args[0], chunk)
    va = _bcast(args[1], chunk)
    return MapColumn(ka.offsets, ka.child, va.child, ka.validity)


def _f_map_keys(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    return c._as_lists()[0]


def _f_map_values(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    return c._as_lists()[1]


def _f_map_contains_key(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    m = _needle_match(c, args[1], chunk).to(torch.int64)
    acc = torch.zeros(len(c), dtype=torch.int64, device=c.device)
    acc.index_add_(0, c.segment_ids(), m)
    return Column(T.BOOL, acc > 0, c.validity)


def _needle_match(c, needle_arg, chunk) -> torch.Tensor:
    """Per-entry bool: key[i] == needle(parent_of_i). Needle may be a
    scalar or a per-row column; keys compared via stable encodings
    (FNV for strings — codes are chunk-local and never comparable)."""
    from .eval import Scalar
    from .joins import fnv_key_tensor, normalize_key

    child = c.keys
    seg = c.segment_ids()
    if isinstance(child, StringColumn):
        ck = fnv_key_tensor(child.decode_dict())
    else:
        ck = normalize_key(child)
    if isinstance(needle_arg, Scalar):
        kl = c._as_lists()[0]
        return _elem_match(kl, needle_arg.value)
    nc = _bcast(needle_arg, chunk)
    if isinstance(nc, StringColumn):
        nk = fnv_key_tensor(nc.decode_dict())
    else:
        nk = normalize_key(nc)
    m = ck == nk.index_select(0, seg)
    return m & child.valid_mask() & nc.valid_mask().index_select(0, seg)


def _f_map_element_at(args, out, chunk, ev):
    """element_at(map, key) / map[key]: value for key, null if absent."""
    c = _bcast(args[0], chunk)
    m = _needle_match(c, args[1], chunk)
    seg = c.segment_ids()
    pos = torch.arange(len(c.keys), dtype=torch.int64, device=c.device)
    cand = torch.where(m, pos, torch.full_like(pos, _BIG))
    first = torch.full((len(c),), _BIG, dtype=torch.int64, device=c.device)
    first.scatter_reduce_(0, seg, cand, reduce="amin", include_self=True)
    found = first != _BIG
    safe = torch.where(found, first, torch.zeros_like(first))
    if len(c.values) == 0:
        return Column.from_values([None] * len(c), c.values.dtype, device=c.device)
    got = c.values.gather(safe.clamp(0, max(len(c.values) - 1, 0)))
    valid = found & got.valid_mask() & c.valid_mask()
    v = None if bool(valid.all()) else valid.to(torch.uint8)
    from .column import StringColumn as _S

    if isinstance(got, _S):
        got.validity = v
        return got
    return Column(got.dtype, got.data, v)


MAP_IMPLS = {
    "map": _f_map, "map_from_arrays": _f_map_from_arrays,
    "map_keys": _f_map_keys, "map_values": _f_map_values,
    "map_contains_key": _f_map_contains_key,
}
IMPLS.update(MAP_IMPLS)


def _f_element_at_dispatch(args, out, chunk, ev):
    """element_at over arrays (1-based) or maps (by key)."""
    from .column import MapColumn

    c = _bcast(args[0], chunk)
    if isinstance(c, MapColumn):
        return _f_map_element_at(args, out, chunk, ev)
    return _f_element_at(args, out, chunk, ev)


def _f_subscript(args, out, chunk, ev):
    """a[i]: 0-based for arrays (Spark), key lookup for maps."""
    from .column import MapColumn

    c = _bcast(args[0], chunk)
    if isinstance(c, MapColumn):
        return _f_map_element_at(args, out, chunk, ev)
    return _f_get(args, out, chunk, ev)


IMPLS["element_at"] = _f_element_at_dispatch
IMPLS["try_element_at"] = _f_element_at_dispatch
IMPLS["element_at_sql"] = _f_subscript


# ---------------------------------------------------------------------------
# struct functions (StructColumn: parallel named children)
# ---------------------------------------------------------------------------

def _f_struct(args, out, chunk, ev):
    from .column import StructColumn

    kids = []
    for f, a in zip(out.fields, args):
        kids.append((f.name, _bcast(a, chunk)))
    return StructColumn(kids, dtype=out)


def _f_named_struct(args, out, chunk, ev):
    from .column import StructColumn

    kids = []
    for i, f in enumerate(out.fields):
        kids.append((f.name, _bcast(args[2 * i + 1], chunk)))
    return StructColumn(kids, dtype=out)


def _f_get_field(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    name = _scalar_value(args[1])
    got = c.field(name)
    if c.validity is not None:
        valid = c.valid_mask() & got.valid_mask()
        v = None if bool(valid.all()) else valid.to(torch.uint8)
        if isinstance(got, (StringColumn, ListColumn)):
            got.validity = v
            return got
        return Column(got.dtype, got.data, v)
    return got


IMPLS["struct"] = _f_struct
IMPLS["named_struct"] = _f_named_struct
IMPLS["get_field"] = _f_get_field


# ---------------------------------------------------------------------------
# higher-order functions: lambda body evaluated ONCE over the flattened
# child (one vectorized pass), params at chunk positions 0..k-1 and the
# enclosing row's columns (repeated per element) at k+i.
# ---------------------------------------------------------------------------

def eval_hof(ev, e, chunk):
    from ..plan import spec as S
    from .chunk import Chunk

    name = e.name.lower()
    if name in ("transform_keys", "transform_values", "map_filter"):
        return eval_map_hof(ev, e, chunk)
    if name == "map_zip_with":
        return _f_map_zip_with_eval(ev, e, chunk)
    if name in ("aggregate", "reduce"):
        return _f_reduce(e.args, e.dtype, chunk, ev)
    if name == "array_sort":
        return _f_array_sort_cmp(ev, e, chunk)
    if name == "zip_with":
        a = _bcast(ev.eval(e.args[0], chunk), chunk)
        b = _bcast(ev.eval(e.args[1], chunk), chunk)
        return _f_zip_with([a, b], e.dtype, chunk, ev, e.args[2])
    arr = _bcast(ev.eval(e.args[0], chunk), chunk)
    lam = e.args[1]
    k = len(lam.params)
    seg = arr.segment_ids()
    total = len(arr.child)
    dev = arr.device
    flat_cols: List = [arr.child]
    if k == 2:
        pos = torch.arange(total, dtype=torch.int64, device=dev) \
            - torch.repeat_interleave(arr.offsets[:-1], arr.lengths())
        flat_cols.append(Column(T.I32, pos.to(torch.int32)))
    for c in chunk.columns:
        flat_cols.append(c.gather(seg) if c is not None else None)
    flat = Chunk(flat_cols, [f"__l{i}" for i in range(len(flat_cols))],
                 chunk.partitioning)
    flat.forced_rows = total
    from .eval import broadcast

    res = broadcast(ev.eval(lam.body, flat), total, dev)
    if name == "transform":
        return ListColumn(arr.offsets, res, arr.validity)
    mask = res.data.to(torch.bool) & res.valid_mask()
    if name == "filter":
        kept = torch.nonzero(mask, as_tuple=False).flatten()
        new_seg = seg.index_select(0, kept)
        lens = torch.bincount(new_seg, minlength=len(arr))
        offs = torch.zeros(len(arr) + 1, dtype=torch.int64, device=dev)
        torch.cumsum(lens, 0, out=offs[1:])
        return ListColumn(offs, arr.child.gather(kept), arr.validity, arr.dtype)
    acc = torch.zeros(len(arr), dtype=torch.int64, device=dev)
    acc.index_add_(0, seg, mask.to(torch.int64))
    if name == "exists":
        return Column(T.BOOL, acc > 0, arr.validity)
    # forall: true when every element satisfies (vacuously true for empty)
    return Column(T.BOOL, acc == arr.lengths(), arr.validity)


def _f_zip_with(args_cols, out, chunk, ev, lam):
    """zip_with(a, b, (x, y) -> e): parallel element walk; shorter side
    null-padded (Spark semantics)."""
    from .chunk import Chunk
    from .eval import broadcast

    a, b = args_cols
    la, lb = a.lengths(), b.lengths()
    lens = torch.maximum(la, lb)
    n = len(a)
    dev = a.device
    offs = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=offs[1:])
    total = int(offs[-1].item())
    seg = torch.repeat_interleave(torch.arange(n, dtype=torch.int64, device=dev), lens)
    pos = torch.arange(total, dtype=torch.int64, device=dev) \
        - torch.repeat_interleave(offs[:-1], lens)

    def padded(c, clens):
        ok = pos < clens.index_select(0, seg)
        idx = (c.offsets[:-1].index_select(0, seg) + pos).clamp(0, max(len(c.child) - 1, 0))
        got = c.child.gather(idx) if len(c.child) else None
        if got is None:
            return Column.from_values([None] * total, c.child.dtype, device=dev)
        valid = ok & got.valid_mask()
        v = None if bool(valid.all()) else valid.to(torch.uint8)
        if isinstance(got, (StringColumn, ListColumn)):
            got.validity = v
            return got
        return Column(got.dtype, got.data, v)

    ea, eb = padded(a, la), padded(b, lb)
    flat_cols = [ea, eb] + [c.gather(seg) if c is not None else None
                            for c in chunk.columns]
    flat = Chunk(flat_cols, [f"__z{i}" for i in range(len(flat_cols))],
                 chunk.partitioning)
    flat.forced_rows = total
    res = broadcast(ev.eval(lam.body, flat), total, dev)
    validity = None
    va, vb = a.valid_mask(), b.valid_mask()
    both = va & vb
    if not bool(both.all()):
        validity = both.to(torch.uint8)
    return ListColumn(offs, res, validity)


def _f_reduce(args, out, chunk, ev):
    """aggregate(arr, init, (acc, x) -> e [, finish]): K vectorized rounds
    (K = max array length) — round k folds element k into the accumulator
    for every row whose array is long enough; no per-row Python loop."""
    from .chunk import Chunk
    from .eval import broadcast, cast_column

    arr = _bcast(ev.eval(args[0], chunk), chunk)
    lam = args[2]
    acc = cast_column(_bcast(ev.eval(args[1], chunk), chunk), lam.body.dtype) \
        if lam.body.dtype is not None else _bcast(ev.eval(args[1], chunk), chunk)
    lens = arr.lengths()
    n = len(arr)
    dev = arr.device
    kmax = int(lens.max().item()) if n else 0
    for k in range(kmax):
        live = lens > k
        idx = (arr.offsets[:-1] + k).clamp(0, max(len(arr.child) - 1, 0))
        elem = arr.child.gather(idx)
        flat_cols = [acc, elem] + list(chunk.columns)
        flat = Chunk(flat_cols, [f"__r{i}" for i in range(len(flat_cols))],
                     chunk.partitioning)
        flat.forced_rows = n
        new_acc = broadcast(ev.eval(lam.body, flat), n, dev)
        # rows whose array ended keep their accumulator
        if bool(live.all()):
            acc = new_acc
        else:
            keep = ~live
            if isinstance(acc, (StringColumn, ListColumn)):
                raise NotImplementedError("aggregate() with string accumulator")
            data = torch.where(keep, acc.data, new_acc.data)
            av = acc.valid_mask()
            nv = new_acc.valid_mask()
            valid = torch.where(keep, av, nv)
            acc = Column(new_acc.dtype, data,
                         None if bool(valid.all()) else valid.to(torch.uint8))
    if len(args) > 3:
        fin = args[3]
        flat = Chunk([acc] + list(chunk.columns),
                     [f"__f{i}" for i in range(1 + len(chunk.columns))],
                     chunk.partitioning)
        flat.forced_rows = n
        acc = broadcast(ev.eval(fin.body, flat), n, dev)
    # null array -> null result
    if arr.validity is not None:
        valid = arr.valid_mask() & acc.valid_mask()
        acc = Column(acc.dtype, acc.data, valid.to(torch.uint8))
    return acc


# ---------------------------------------------------------------------------
# array set ops / append / repeat / flatten
# ---------------------------------------------------------------------------

def _concat_rows(a: ListColumn, b: ListColumn) -> ListColumn:
    """Per-row concatenation: out[i] = a[i] ++ b[i]."""
    from .executor import concat_columns

    n = len(a)
    dev = a.device
    la, lb = a.lengths(), b.lengths()
    lens = la + lb
    offs = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=offs[1:])
    total = int(offs[-1].item())
    seg = torch.repeat_interleave(torch.arange(n, dtype=torch.int64, device=dev), lens)
    pos = torch.arange(total, dtype=torch.int64, device=dev) \
        - torch.repeat_interleave(offs[:-1], lens)
    from_a = pos < la.index_select(0, seg)
    na = len(a.child)
    idx_a = (a.offsets[:-1].index_select(0, seg) + pos).clamp(0, max(na - 1, 0))
    idx_b = (b.offsets[:-1].index_select(0, seg)
             + (pos - la.index_select(0, seg))).clamp(0, max(len(b.child) - 1, 0))
    # gather from the concatenation of both children with one index space
    allc = concat_columns([a.child, b.child]) if len(b.child) or len(a.child) else a.child
    idx = torch.where(from_a, idx_a, idx_b + na)
    child = allc.gather(idx) if len(allc) else a.child
    validity = None
    va, vb = a.valid_mask(), b.valid_mask()
    both = va & vb
    if not bool(both.all()):
        validity = both.to(torch.uint8)
    return ListColumn(offs, child, validity)


def _f_array_union(args, out, chunk, ev):
    a = _bcast(args[0], chunk)
    b = _bcast(args[1], chunk)
    cat = _concat_rows(a, b)
    return _f_array_distinct([cat], out, chunk, ev)


def _row_membership(a: ListColumn, b: ListColumn) -> torch.Tensor:
    """Per-element-of-a bool: element also present in b's SAME row."""
    from .joins import fnv_key_tensor, normalize_key

    def key(ch):
        if isinstance(ch, StringColumn):
            return fnv_key_tensor(ch.decode_dict())
        return normalize_key(ch)

    ka, kb = key(a.child), key(b.child)
    sa, sb = a.segment_ids(), b.segment_ids()
    pa = _mix_rowkey(sa, ka)
    pb = _mix_rowkey(sb, kb)
    if pb.numel() == 0:
        return torch.zeros(pa.shape[0], dtype=torch.bool, device=a.device)
    sorted_b = torch.sort(pb).values
    idx = torch.searchsorted(sorted_b, pa).clamp_max(sorted_b.shape[0] - 1)
    return sorted_b.index_select(0, idx) == pa


def _mix_rowkey(seg: torch.Tensor, key: torch.Tensor) -> torch.Tensor:
    from ..exec.distributed import _mix64

    return _mix64(seg * 31 + _mix64(key))


def _filter_elements(c: ListColumn, keep: torch.Tensor) -> ListColumn:
    kept = torch.nonzero(keep, as_tuple=False).flatten()
    seg = c.segment_ids().index_select(0, kept)
    lens = torch.bincount(seg, minlength=len(c))
    offs = torch.zeros(len(c) + 1, dtype=torch.int64, device=c.device)
    torch.cumsum(lens, 0, out=offs[1:])
    return ListColumn(offs, c.child.gather(kept), c.validity, c.dtype)


def _f_array_intersect(args, out, chunk, ev):
    a = _bcast(args[0], chunk)
    b = _bcast(args[1], chunk)
    da = _f_array_distinct([a], out, chunk, ev)
    return _filter_elements(da, _row_membership(da, b))


def _f_array_except(args, out, chunk, ev):
    a = _bcast(args[0], chunk)
    b = _bcast(args[1], chunk)
    da = _f_array_distinct([a], out, chunk, ev)
    return _filter_elements(da, ~_row_membership(da, b))


def _f_array_remove(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    m = _elem_match(c, _scalar_value(args[1]))
    return _filter_elements(c, ~m)


def _f_array_compact(args, out, chunk, ev):
    c = _bcast(args[0], chunk)
    return _filter_elements(c, c.child.valid_mask())


def _f_flatten(args, out, chunk, ev):
    c = _bcast(args[0], chunk)  # list<list<T>>
    inner = c.child
    if not isinstance(inner, ListColumn):
        raise NotImplementedError("flatten expects array<array<...>>")
    if inner.validity is not None:
        raise NotImplementedError("flatten with null inner arrays")
    # inner lists are laid out contiguously: compose offsets directly
    new_offs = inner.offsets.index_select(0, c.offsets)
    return ListColumn(new_offs, inner.child, c.validity)


def _f_array_repeat(args, out, chunk, ev):
    from .eval import broadcast

    n = chunk.num_rows
    dev = chunk.device
    v = broadcast(args[0], n, dev)
    cnt = _bcast(args[1], chunk).data.to(torch.int64).clamp_min(0)
    offs = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(cnt, 0, out=offs[1:])
    seg = torch.repeat_interleave(torch.arange(n, dtype=torch.int64, device=dev), cnt)
    child = v.gather(seg)
    return ListColumn(offs, child)


def _f_array_append(args, out, chunk, ev):
    from .eval import broadcast, cast_column

    a = _bcast(args[0], chunk)
    v = cast_column(broadcast(args[1], chunk.num_rows, chunk.device),
                    a.child.dtype)
    one = torch.ones(len(a), dtype=torch.int64, device=a.device)
    offs = torch.arange(0, len(a) + 1, dtype=torch.int64, device=a.device)
    single = ListColumn(offs, v, None)
    return _concat_rows(a, single)


def _f_array_prepend(args, out, chunk, ev):
    from .eval import broadcast, cast_column

    a = _bcast(args[0], chunk)
    v = cast_column(broadcast(args[1], chunk.num_rows, chunk.device),
                    a.child.dtype)
    offs = torch.arange(0, len(a) + 1, dtype=torch.int64, device=a.device)
    single = ListColumn(offs, v, None)
    return _concat_rows(single, a)


IMPLS.update({
    "array_union": _f_array_union, "array_intersect": _f_array_intersect,
    "array_except": _f_array_except, "array_remove": _f_array_remove,
    "array_compact": _f_array_compact, "flatten": _f_flatten,
    "array_repeat": _f_array_repeat, "array_append": _f_array_append,
    "array_prepend": _f_array_prepend,
})


def eval_map_hof(ev, e, chunk):
    """transform_keys/transform_values/map_filter: lambda body over the flat
    (key, value) entry columns — params at 0,1, enclosing row columns at 2+."""
    from .chunk import Chunk
    from .column import MapColumn
    from .eval import broadcast

    name = e.name.lower()
    m = _bcast(ev.eval(e.args[0], chunk), chunk)
    lam = e.args[1]
    seg = m.segment_ids()
    total = len(m.keys)
    dev = m.device
    flat_cols = [m.keys, m.values] + [c.gather(seg) if c is not None else None
                                      for c in chunk.columns]
    flat = Chunk(flat_cols, [f"__m{i}" for i in range(len(flat_cols))],
                 chunk.partitioning)
    flat.forced_rows = total
    res = broadcast(ev.eval(lam.body, flat), total, dev)
    if name == "transform_keys":
        return MapColumn(m.offsets, res, m.values, m.validity)
    if name == "transform_values":
        return MapColumn(m.offsets, m.keys, res, m.validity)
    # map_filter
    keep = res.data.to(torch.bool) & res.valid_mask()
    kept = torch.nonzero(keep, as_tuple=False).flatten()
    new_seg = seg.index_select(0, kept)
    lens = torch.bincount(new_seg, minlength=len(m))
    offs = torch.zeros(len(m) + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=offs[1:])
    return MapColumn(offs, m.keys.gather(kept), m.values.gather(kept),
                     m.validity)


# ===========================================================================
# structural array/map builders (host path — these are shape-changing,
# allocation-bound ops, not hot compute; ref: sail-function scalar/map,
# scalar/collection)
# ===========================================================================
def _col(a, chunk):
    from .eval import broadcast

    return broadcast(a, chunk.num_rows, chunk.device)


def _scalarize(a):
    from .functions_impl import _scalarize as _sc

    return _sc(a)


def _build_list_generic(rows, elem_type, device):
    """ListColumn from python lists whose element type may be a struct."""
    from .column import MapColumn, StructColumn

    if isinstance(elem_type, T.StructType):
        flat = [x for r in rows if r is not None for x in r]
        offs = [0]
        for r in rows:
            offs.append(offs[-1] + (len(r) if r is not None else 0))
        kids = []
        for i, f in enumerate(elem_type.fields):
            vals = [None if x is None else
                    (x.get(f.name) if isinstance(x, dict) else x[i])
                    for x in flat]
            if isinstance(f.dtype, T.StringType):
                kids.append((f.name, StringColumn.from_pylist(
                    vals, device=str(device))))
            else:
                kids.append((f.name, Column.from_values(
                    vals, f.dtype, device=device)))
        child = StructColumn(kids, dtype=elem_type)
        validity = None
        if any(r is None for r in rows):
            validity = torch.tensor([0 if r is None else 1 for r in rows],
                                    dtype=torch.uint8, device=device)
        return ListColumn(torch.tensor(offs, dtype=torch.int64, device=device),
                          child, validity,
                          T.ArrayType(elem_type))
    return ListColumn.from_pylist(rows, elem_type, device=str(device))


def _f_arrays_zip(args, out, chunk, ev):
    """arrays_zip(a1, a2, ...) -> array<struct> (shorter arrays padded with
    nulls, Spark semantics)."""
    cols = [_col(a, chunk) for a in args]
    pys = [c.to_pylist() for c in cols]
    n = chunk.num_rows
    rows = []
    for i in range(n):
        arrs = [p[i] for p in pys]
        if all(a is None for a in arrs):
            rows.append(None)
            continue
        ln = max(len(a) for a in arrs if a is not None)
        rows.append([tuple((a[j] if a is not None and j < len(a) else None)
                           for a in arrs) for j in range(ln)])
    return _build_list_generic(rows, out.element, chunk.device)


def _f_array_insert(args, out, chunk, ev):
    """array_insert(arr, pos, val): 1-based; negative counts from the end."""
    arr = _col(args[0], chunk)
    rows = arr.to_pylist()
    from .eval import broadcast

    n = chunk.num_rows
    pos = broadcast(args[1], n, chunk.device).to_pylist()
    val = broadcast(args[2], n, chunk.device).to_pylist()
    res = []
    for r, p, v in zip(rows, pos, val):
        if r is None or p is None or p == 0:
            res.append(None)
            continue
        r = list(r)
        if p > 0:
            while len(r) < p - 1:
                r.append(None)
            r.insert(p - 1, v)
        else:
            k = len(r) + p + 1
            if k < 0:
                r = [v] + [None] * (-k) + r
            else:
                r.insert(k, v)
        res.append(r)
    return _build_list_generic(res, out.element, chunk.device)


def _f_array_contains_all(args, out, chunk, ev):
    a = _col(args[0], chunk).to_pylist()
    b = _col(args[1], chunk).to_pylist()
    vals = [None if (x is None or y is None)
            else all(e in x for e in y) for x, y in zip(a, b)]
    return Column.from_values(vals, T.BOOL, device=chunk.device)


def _f_map_concat(args, out, chunk, ev):
    maps = [_col(a, chunk).to_pylist() for a in args]
    n = chunk.num_rows
    rows = []
    for i in range(n):
        ms = [m[i] for m in maps]
        if all(m is None for m in ms):
            rows.append(None)
            continue
        merged = {}
        for m in ms:
            if m:
                merged.update(m)  # later maps win (Spark LAST_WIN policy)
        rows.append(merged)
    from .column import MapColumn

    return MapColumn.from_pylist(rows, out.key, out.value,
                                 device=str(chunk.device))


def _f_map_entries(args, out, chunk, ev):
    m = _col(args[0], chunk).to_pylist()
    rows = [None if r is None else [(k, v) for k, v in r.items()] for r in m]
    return _build_list_generic(rows, out.element, chunk.device)


def _f_map_from_entries(args, out, chunk, ev):
    rows = _col(args[0], chunk).to_pylist()
    res = []
    for r in rows:
        if r is None:
            res.append(None)
            continue
        d = {}
        for e in r:
            if e is None:
                continue
            k, v = (e.get(f.name) for f in args[0].dtype.element.fields) \
                if isinstance(e, dict) else (e[0], e[1])
            d[k] = v
        res.append(d)
    from .column import MapColumn

    return MapColumn.from_pylist(res, out.key, out.value,
                                 device=str(chunk.device))


def _f_str_to_map(args, out, chunk, ev):
    c = _col(args[0], chunk)
    pair_d = _scalarize(args[1]).value if len(args) > 1 else ","
    kv_d = _scalarize(args[2]).value if len(args) > 2 else ":"
    rows = []
    for v in c.to_pylist():
        if v is None:
            rows.append(None)
            continue
        d = {}
        for pair in v.split(pair_d):
            if kv_d in pair:
                k, val = pair.split(kv_d, 1)
                d[k] = val
            elif pair:
                d[pair] = None
        rows.append(d)
    from .column import MapColumn

    return MapColumn.from_pylist(rows, T.STRING, T.STRING,
                                 device=str(chunk.device))


def _col_from_py(vals, dtype, device):
    if isinstance(dtype, T.StringType):
        return StringColumn.from_pylist(vals, device=str(device))
    return Column.from_values(vals, dtype, device=device)


def _f_map_zip_with_eval(ev, e, chunk):
    """map_zip_with(m1, m2, (k, v1, v2) -> ...): per-row key union, lambda
    over the aligned flat entries (params at 0..2, row columns at 3+)."""
    from .chunk import Chunk
    from .column import MapColumn
    from .eval import broadcast

    m1 = _bcast(ev.eval(e.args[0], chunk), chunk)
    m2 = _bcast(ev.eval(e.args[1], chunk), chunk)
    lam = e.args[2]
    r1, r2 = m1.to_pylist(), m2.to_pylist()
    n = chunk.num_rows
    keys, v1, v2, offs, valid = [], [], [], [0], []
    for i in range(n):
        a, b = r1[i], r2[i]
        if a is None and b is None:
            valid.append(0)
            offs.append(offs[-1])
            continue
        valid.append(1)
        a = a or {}
        b = b or {}
        uk = list(a.keys()) + [k for k in b.keys() if k not in a]
        keys.extend(uk)
        v1.extend(a.get(k) for k in uk)
        v2.extend(b.get(k) for k in uk)
        offs.append(offs[-1] + len(uk))
    dev = chunk.device
    kcol = _col_from_py(keys, m1.dtype.key, dev)
    flat_cols = [kcol,
                 _col_from_py(v1, m1.dtype.value, dev),
                 _col_from_py(v2, m2.dtype.value, dev)]
    seg_rows = []
    for i in range(n):
        seg_rows.extend([i] * (offs[i + 1] - offs[i]))
    seg = torch.tensor(seg_rows, dtype=torch.int64, device=dev)
    flat_cols += [c.gather(seg) if c is not None else None
                  for c in chunk.columns]
    flat = Chunk(flat_cols, [f"__z{i}" for i in range(len(flat_cols))],
                 chunk.partitioning)
    flat.forced_rows = len(keys)
    res = broadcast(ev.eval(lam.body, flat), len(keys), dev)
    validity = None
    if not all(valid):
        validity = torch.tensor(valid, dtype=torch.uint8, device=dev)
    return MapColumn(torch.tensor(offs, dtype=torch.int64, device=dev),
                     kcol, res, validity, dtype=e.dtype)


# -- vector math over array<double> (segment ops, device path) -------------
def _vec(args, chunk, i=0):
    c = _col(args[i], chunk)
    return c


def _f_vector_norm(args, out, chunk, ev):
    c = _vec(args, chunk)
    p = float(_scalarize(args[1]).value) if len(args) > 1 else 2.0
    seg = c.segment_ids()
    x = c.child.data.to(torch.float64).abs() ** p
    sums = torch.zeros(len(c), dtype=torch.float64, device=c.device)
    sums.index_add_(0, seg, x)
    return Column(T.F64, sums ** (1.0 / p), c.validity)


def _f_vector_inner_product(args, out, chunk, ev):
    a, b = _vec(args, chunk, 0), _vec(args, chunk, 1)
    seg = a.segment_ids()
    prod = a.child.data.to(torch.float64) * b.child.data.to(torch.float64)
    sums = torch.zeros(len(a), dtype=torch.float64, device=a.device)
    sums.index_add_(0, seg, prod)
    return Column(T.F64, sums, a.validity)


def _f_vector_l2_distance(args, out, chunk, ev):
    a, b = _vec(args, chunk, 0), _vec(args, chunk, 1)
    seg = a.segment_ids()
    d = a.child.data.to(torch.float64) - b.child.data.to(torch.float64)
    sums = torch.zeros(len(a), dtype=torch.float64, device=a.device)
    sums.index_add_(0, seg, d * d)
    return Column(T.F64, sums.sqrt(), a.validity)


def _f_vector_normalize(args, out, chunk, ev):
    c = _vec(args, chunk)
    seg = c.segment_ids()
    x = c.child.data.to(torch.float64)
    sums = torch.zeros(len(c), dtype=torch.float64, device=c.device)
    sums.index_add_(0, seg, x * x)
    norm = sums.sqrt().clamp_min(1e-300)
    return ListColumn(c.offsets, Column(T.F64, x / norm[seg], None),
                      c.validity, T.ArrayType(T.F64))


IMPLS["arrays_zip"] = _f_arrays_zip
IMPLS["array_insert"] = _f_array_insert
IMPLS["array_contains_all"] = _f_array_contains_all
IMPLS["map_concat"] = _f_map_concat
IMPLS["map_entries"] = _f_map_entries
IMPLS["map_from_entries"] = _f_map_from_entries
IMPLS["str_to_map"] = _f_str_to_map
IMPLS["vector_norm"] = _f_vector_norm
IMPLS["vector_inner_product"] = _f_vector_inner_product
IMPLS["vector_l2_distance"] = _f_vector_l2_distance
IMPLS["vector_normalize"] = _f_vector_normalize

def _f_array_sort_cmp(ev, e, chunk):
    """array_sort(arr, (a, b) -> cmp): comparator sort. Evaluates the
    lambda once over every within-row element PAIR (vectorized), ranks
    each element by its number of wins — valid for any consistent
    comparator — then permutes children per row. Pair count is m^2 per
    row; arrays are small, guarded by a cap."""
    from .chunk import Chunk as _Ck

    arr = _bcast(ev.eval(e.args[0], chunk), chunk)
    lam = e.args[1]
    n = len(arr)
    dev = arr.device
    lens = arr.lengths()
    m2 = lens * lens
    total_pairs = int(m2.sum().item())
    if total_pairs > 50_000_000:
        raise ValueError("array_sort comparator: arrays too large")
    if total_pairs == 0:
        return arr
    row = torch.repeat_interleave(torch.arange(n, device=dev), m2)
    starts2 = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(m2, 0, out=starts2[1:])
    p = torch.arange(total_pairs, device=dev) \
        - starts2[:-1].index_select(0, row)
    mrow = lens.index_select(0, row).clamp_min(1)
    i = p // mrow
    j = p % mrow
    base = arr.offsets[:-1].index_select(0, row)
    xi = arr.child.gather(base + i)
    xj = arr.child.gather(base + j)
    outer_cols = [c.gather(row) for c in chunk.columns]
    sub = _Ck([xi, xj] + outer_cols,
              ["__p0", "__p1"] + list(chunk.names))
    res = _bcast(ev.eval(lam.body, sub), sub)
    r = res.data.to(torch.int64)
    r = torch.where(res.valid_mask(), r, torch.zeros_like(r))
    nchild = len(arr.child)
    wins = torch.zeros(nchild, dtype=torch.int64, device=dev)
    wins.index_add_(0, base + i, (r > 0).to(torch.int64))
    local = torch.arange(nchild, device=dev) \
        - torch.repeat_interleave(arr.offsets[:-1], lens)
    key = wins * (nchild + 1) + local
    seg = arr.segment_ids()
    big = int(key.max().item()) + 2
    perm = torch.argsort(seg * big + key, stable=True)
    child_sorted = arr.child.gather(perm)
    return ListColumn(arr.offsets, child_sorted, arr.validity, arr.dtype)
