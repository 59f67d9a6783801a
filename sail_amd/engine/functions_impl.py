"""Scalar function implementations (torch path).

The dispatch table of per-function kernels used by the Evaluator; Spark
semantics (ref: crates/sail-function/src/scalar/* for behavior parity).
Functions receive `Val`s (Column or Scalar) and the enclosing chunk.
"""
from __future__ import annotations

import datetime as _dt
import math
from typing import List

import torch

from . import types as T
from .chunk import Chunk
from .column import Column, StringColumn

_EPOCH = _dt.date(1970, 1, 1)


def dispatch_function(name: str, args: List, out_type, chunk: Chunk, ev):
    fn = _IMPLS.get(name)
    if fn is None:
        raise NotImplementedError(f"function {name} has no implementation yet")
    return fn(args, out_type, chunk, ev)


def _col(v, chunk: Chunk) -> Column:
    from .eval import broadcast

    return broadcast(v, chunk.num_rows, chunk.device)


def _scalarize(v):
    from .eval import Scalar

    return v if isinstance(v, Scalar) else None


# ---------------------------------------------------------------------------
# datetime — civil-calendar math on int32 day counts, vectorized.
# Uses the standard days-from-civil algorithm (Howard Hinnant's) so that
# year/month extraction runs as pure tensor arithmetic on device.
# ---------------------------------------------------------------------------

def _civil_from_days(z: torch.Tensor):
    z = z.to(torch.int64) + 719468
    era = torch.div(torch.where(z >= 0, z, z - 146096), 146097, rounding_mode="floor")
    doe = z - era * 146097
    yoe = torch.div(doe - torch.div(doe, 1460, rounding_mode="floor")
                    + torch.div(doe, 36524, rounding_mode="floor")
                    - torch.div(doe, 146096, rounding_mode="floor"), 365, rounding_mode="floor")
    y = yoe + era * 400
    doy = doe - (365 * yoe + torch.div(yoe, 4, rounding_mode="floor") - torch.div(yoe, 100, rounding_mode="floor"))
    mp = torch.div(5 * doy + 2, 153, rounding_mode="floor")
    d = doy - torch.div(153 * mp + 2, 5, rounding_mode="floor") + 1
    m = mp + torch.where(mp < 10, torch.full_like(mp, 3), torch.full_like(mp, -9))
    y = y + (m <= 2).to(torch.int64)
    return y, m, d


def _days_from_civil(y: torch.Tensor, m: torch.Tensor, d: torch.Tensor) -> torch.Tensor:
    y = y - (m <= 2).to(torch.int64)
    era = torch.div(torch.where(y >= 0, y, y - 399), 400, rounding_mode="floor")
    yoe = y - era * 400
    mp = torch.where(m > 2, m - 3, m + 9)
    doy = torch.div(153 * mp + 2, 5, rounding_mode="floor") + d - 1
    doe = yoe * 365 + torch.div(yoe, 4, rounding_mode="floor") - torch.div(yoe, 100, rounding_mode="floor") + doy
    return era * 146097 + doe - 719468


def _as_days(c: Column) -> torch.Tensor:
    if isinstance(c.dtype, T.TimestampType):
        return torch.div(c.data, 86_400_000_000, rounding_mode="floor")
    return c.data.to(torch.int64)


def _f_year(args, out, chunk, ev):
    c = _col(args[0], chunk)
    y, m, d = _civil_from_days(_as_days(c))
    return Column(T.I32, y.to(torch.int32), c.validity)


def _f_month(args, out, chunk, ev):
    c = _col(args[0], chunk)
    y, m, d = _civil_from_days(_as_days(c))
    return Column(T.I32, m.to(torch.int32), c.validity)


def _f_day(args, out, chunk, ev):
    c = _col(args[0], chunk)
    y, m, d = _civil_from_days(_as_days(c))
    return Column(T.I32, d.to(torch.int32), c.validity)


def _f_quarter(args, out, chunk, ev):
    c = _col(args[0], chunk)
    y, m, d = _civil_from_days(_as_days(c))
    return Column(T.I32, (torch.div(m - 1, 3, rounding_mode="floor") + 1).to(torch.int32), c.validity)


def _f_dayofweek(args, out, chunk, ev):
    # Spark: 1 = Sunday ... 7 = Saturday
    c = _col(args[0], chunk)
    days = _as_days(c)
    dow = torch.remainder(days + 4, 7) + 1  # 1970-01-01 was Thursday
    return Column(T.I32, dow.to(torch.int32), c.validity)


def _f_weekday(args, out, chunk, ev):
    # Spark weekday: 0 = Monday ... 6 = Sunday
    c = _col(args[0], chunk)
    days = _as_days(c)
    wd = torch.remainder(days + 3, 7)
    return Column(T.I32, wd.to(torch.int32), c.validity)


def _f_dayofyear(args, out, chunk, ev):
    c = _col(args[0], chunk)
    days = _as_days(c)
    y, m, d = _civil_from_days(days)
    jan1 = _days_from_civil(y, torch.ones_like(y), torch.ones_like(y))
    return Column(T.I32, (days - jan1 + 1).to(torch.int32), c.validity)


def _f_date_add(args, out, chunk, ev):
    c = _col(args[0], chunk)
    n = _col(args[1], chunk)
    return Column(T.DATE, (_as_days(c) + n.data.to(torch.int64)).to(torch.int32),
                  _merge(c, n))


def _f_date_sub(args, out, chunk, ev):
    c = _col(args[0], chunk)
    n = _col(args[1], chunk)
    return Column(T.DATE, (_as_days(c) - n.data.to(torch.int64)).to(torch.int32),
                  _merge(c, n))


def _f_datediff(args, out, chunk, ev):
    a = _col(args[0], chunk)
    b = _col(args[1], chunk)
    return Column(T.I32, (_as_days(a) - _as_days(b)).to(torch.int32), _merge(a, b))


def _f_add_months(args, out, chunk, ev):
    c = _col(args[0], chunk)
    n = _col(args[1], chunk)
    y, m, d = _civil_from_days(_as_days(c))
    total = y * 12 + (m - 1) + n.data.to(torch.int64)
    ny = torch.div(total, 12, rounding_mode="floor")
    nm = total - ny * 12 + 1
    # clamp day to end of month
    dim = _days_in_month(ny, nm)
    nd = torch.minimum(d, dim)
    return Column(T.DATE, _days_from_civil(ny, nm, nd).to(torch.int32), _merge(c, n))


def _days_in_month(y: torch.Tensor, m: torch.Tensor) -> torch.Tensor:
    base = torch.tensor([31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31],
                        dtype=torch.int64, device=y.device)
    dim = base[(m - 1).clamp(0, 11)]
    leap = ((y % 4 == 0) & ((y % 100 != 0) | (y % 400 == 0))) & (m == 2)
    return dim + leap.to(torch.int64)


def _f_last_day(args, out, chunk, ev):
    c = _col(args[0], chunk)
    y, m, d = _civil_from_days(_as_days(c))
    dim = _days_in_month(y, m)
    return Column(T.DATE, _days_from_civil(y, m, dim).to(torch.int32), c.validity)


def _f_trunc(args, out, chunk, ev):
    c = _col(args[0], chunk)
    fmt = _scalarize(args[1])
    unit = (fmt.value if fmt else "month").lower()
    y, m, d = _civil_from_days(_as_days(c))
    one = torch.ones_like(y)
    if unit in ("year", "yyyy", "yy"):
        days = _days_from_civil(y, one, one)
    elif unit in ("quarter",):
        qm = (torch.div(m - 1, 3, rounding_mode="floor")) * 3 + 1
        days = _days_from_civil(y, qm, one)
    elif unit in ("month", "mm", "mon"):
        days = _days_from_civil(y, m, one)
    elif unit in ("week",):
        wd = torch.remainder(_as_days(c) + 3, 7)
        days = _as_days(c) - wd
    else:
        days = _as_days(c)
    return Column(T.DATE, days.to(torch.int32), c.validity)


def _f_date_trunc(args, out, chunk, ev):
    # date_trunc(fmt, timestamp) -> timestamp
    fmt = _scalarize(args[0])
    c = _col(args[1], chunk)
    unit = (fmt.value if fmt else "day").lower()
    us = c.data.to(torch.int64)
    day_us = 86_400_000_000
    if unit in ("year", "yyyy", "yy", "quarter", "month", "mm", "mon", "week"):
        days = torch.div(us, day_us, rounding_mode="floor")
        dcol = Column(T.DATE, days.to(torch.int32), c.validity)
        tr = _f_trunc([dcol, args[0]], T.DATE, chunk, ev)
        return Column(T.TIMESTAMP, tr.data.to(torch.int64) * day_us, c.validity)
    step = {"day": day_us, "dd": day_us, "hour": 3_600_000_000,
            "minute": 60_000_000, "second": 1_000_000}.get(unit, day_us)
    return Column(T.TIMESTAMP, torch.div(us, step, rounding_mode="floor") * step, c.validity)


def _f_to_date(args, out, chunk, ev):
    from .eval import cast_value

    return cast_value(args[0], T.DATE, chunk)


def _f_make_date(args, out, chunk, ev):
    y = _col(args[0], chunk).data.to(torch.int64)
    m = _col(args[1], chunk).data.to(torch.int64)
    d = _col(args[2], chunk).data.to(torch.int64)
    return Column(T.DATE, _days_from_civil(y, m, d).to(torch.int32), None)


def _f_hour(args, out, chunk, ev):
    c = _col(args[0], chunk)
    us = torch.remainder(c.data.to(torch.int64), 86_400_000_000)
    return Column(T.I32, torch.div(us, 3_600_000_000, rounding_mode="floor").to(torch.int32), c.validity)


def _f_minute(args, out, chunk, ev):
    c = _col(args[0], chunk)
    us = torch.remainder(c.data.to(torch.int64), 3_600_000_000)
    return Column(T.I32, torch.div(us, 60_000_000, rounding_mode="floor").to(torch.int32), c.validity)


def _f_second(args, out, chunk, ev):
    c = _col(args[0], chunk)
    us = torch.remainder(c.data.to(torch.int64), 60_000_000)
    return Column(T.I32, torch.div(us, 1_000_000, rounding_mode="floor").to(torch.int32), c.validity)


# ---------------------------------------------------------------------------
# math
# ---------------------------------------------------------------------------

def _merge(a: Column, b: Column):
    from .eval import _merge_validity

    return _merge_validity(a, b)


def _f_abs(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return Column(c.dtype, torch.abs(c.data), c.validity)


def _unary_float(fn):
    def impl(args, out, chunk, ev):
        c = _col(args[0], chunk)
        x = c.data.to(torch.float64)
        if isinstance(c.dtype, T.DecimalType):
            x = x / (10.0 ** c.dtype.scale)
        return Column(T.F64, fn(x), c.validity)

    return impl


def _f_round(args, out, chunk, ev):
    c = _col(args[0], chunk)
    nd = _scalarize(args[1]).value if len(args) > 1 else 0
    if isinstance(c.dtype, T.DecimalType):
        from .eval import _rescale_int

        s = c.dtype.scale
        if nd >= s:
            return c
        data = _rescale_int(c.data, s, nd)
        return Column(T.DecimalType(c.dtype.precision, nd), data, c.validity)
    if c.dtype.is_integer:
        return c
    f = 10.0 ** nd
    x = c.data.to(torch.float64) * f
    # HALF_UP (away from zero), like Spark round
    data = torch.where(x >= 0, torch.floor(x + 0.5), torch.ceil(x - 0.5)) / f
    return Column(T.F64, data, c.validity)


def _f_floor(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if isinstance(c.dtype, T.DecimalType):
        s = 10 ** c.dtype.scale
        data = torch.div(torch.where(c.data >= 0, c.data, c.data - (s - 1)), s, rounding_mode="trunc")
        return Column(out, data, c.validity)
    if c.dtype.is_integer:
        return Column(T.I64, c.data.to(torch.int64), c.validity)
    return Column(T.I64, torch.floor(c.data.to(torch.float64)).to(torch.int64), c.validity)


def _f_ceil(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if isinstance(c.dtype, T.DecimalType):
        s = 10 ** c.dtype.scale
        data = torch.div(torch.where(c.data >= 0, c.data + (s - 1), c.data), s, rounding_mode="trunc")
        return Column(out, data, c.validity)
    if c.dtype.is_integer:
        return Column(T.I64, c.data.to(torch.int64), c.validity)
    return Column(T.I64, torch.ceil(c.data.to(torch.float64)).to(torch.int64), c.validity)


def _f_power(args, out, chunk, ev):
    a = _col(args[0], chunk)
    b = _col(args[1], chunk)
    return Column(T.F64, torch.pow(a.data.to(torch.float64), b.data.to(torch.float64)), _merge(a, b))


def _minmax_n(args, out, chunk, ev, is_max: bool):
    """greatest/least skip NULL arguments; all-NULL rows are NULL
    (Spark semantics)."""
    from .eval import cast_value

    cols = [_col(cast_value(a, out, chunk), chunk) for a in args]
    data = cols[0].data.clone()
    any_valid = cols[0].valid_mask().clone()
    for c in cols[1:]:
        cv = c.valid_mask()
        cd = c.data.to(data.dtype)
        op = torch.maximum if is_max else torch.minimum
        cand = op(data, cd)
        # rows where only one side is valid take that side
        data = torch.where(any_valid & cv, cand,
                           torch.where(cv, cd, data))
        any_valid = any_valid | cv
    return Column(out, data,
                  None if bool(any_valid.all())
                  else any_valid.to(torch.uint8))


def _f_greatest(args, out, chunk, ev):
    return _minmax_n(args, out, chunk, ev, True)


def _f_least(args, out, chunk, ev):
    return _minmax_n(args, out, chunk, ev, False)


def _f_sign(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return Column(T.F64, torch.sign(c.data.to(torch.float64)), c.validity)


def _f_isnan(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if c.dtype.is_float:
        return Column(T.BOOL, torch.isnan(c.data), None)
    return Column(T.BOOL, torch.zeros(len(c), dtype=torch.bool, device=c.device), None)


# ---------------------------------------------------------------------------
# conditional
# ---------------------------------------------------------------------------

def _f_coalesce(args, out, chunk, ev):
    from .eval import Scalar, cast_value

    cols = []
    for a in args:
        if isinstance(a, Scalar) and a.is_null:
            continue
        cols.append(cast_value(a, out, chunk))
    if not cols:
        return Scalar(None, out)
    if isinstance(out, T.StringType):
        # pick-per-row across string columns: codes are chunk-local, so
        # select via a union concat + gather (device-native)
        from .column import StringColumn
        from .executor import _concat_strings

        scols = [_col(c, chunk) for c in cols]
        n = chunk.num_rows
        dev = chunk.device
        sel = torch.full((n,), -1, dtype=torch.int64, device=dev)
        for j, c in enumerate(scols):
            sel = torch.where((sel < 0) & c.valid_mask(),
                              torch.full_like(sel, j), sel)
        valid = sel >= 0
        allc = _concat_strings(scols)
        idx = sel.clamp_min(0) * n + torch.arange(n, device=dev)
        got = allc.gather(idx)
        got.validity = None if bool(valid.all()) else valid.to(torch.uint8)
        return got
    first = _col(cols[0], chunk)
    data = first.data.clone()
    valid = first.valid_mask().clone()
    for a in cols[1:]:
        c = _col(a, chunk)
        need = ~valid
        data = torch.where(need, c.data.to(data.dtype), data)
        valid = valid | (need & c.valid_mask())
    return Column(out, data, None if bool(valid.all()) else valid.to(torch.uint8))


def _f_if(args, out, chunk, ev):
    from .eval import Scalar, cast_value

    cond = _col(args[0], chunk)
    a = _col(cast_value(args[1], out, chunk), chunk)
    b = _col(cast_value(args[2], out, chunk), chunk)
    cmask = cond.data.to(torch.bool) & cond.valid_mask()
    data = torch.where(cmask, a.data, b.data.to(a.data.dtype))
    valid = torch.where(cmask, a.valid_mask(), b.valid_mask())
    return Column(out, data, None if bool(valid.all()) else valid.to(torch.uint8))


def _f_nullif(args, out, chunk, ev):
    a = _col(args[0], chunk)
    b = _col(args[1], chunk)
    eq = a.data == b.data.to(a.data.dtype)
    valid = a.valid_mask() & ~eq
    return Column(out, a.data, valid.to(torch.uint8))


# ---------------------------------------------------------------------------
# strings — CPU reference implementations round-trip through host python;
# the GPU path uses dict-encoding (O(|dict|) host work) or HIP string kernels.
# ---------------------------------------------------------------------------

def _str_map(c: StringColumn, fn, out_is_string=True):
    vals = c.to_pylist()
    res = [None if v is None else fn(v) for v in vals]
    if out_is_string:
        return StringColumn.from_pylist(res, device=c.device)
    return res


def _f_upper(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if isinstance(c, StringColumn) and c.is_dict:
        vals = [v.upper() for v in c.dict_values()]
        from .column import _pack_strings

        offs, byts = _pack_strings(vals, c.device)
        return StringColumn(offs, byts, c.validity, c.codes)
    return _str_map(c, str.upper)


def _f_lower(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if isinstance(c, StringColumn) and c.is_dict:
        vals = [v.lower() for v in c.dict_values()]
        from .column import _pack_strings

        offs, byts = _pack_strings(vals, c.device)
        return StringColumn(offs, byts, c.validity, c.codes)
    return _str_map(c, str.lower)


def _f_octet_length(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if not isinstance(c, StringColumn):
        raise NotImplementedError("octet_length on non-string")
    lens = (c.offsets[1:] - c.offsets[:-1]).to(torch.int32)
    if c.is_dict:
        return Column(T.I32, lens[c.codes.long().clamp_min(0)], c.validity)
    return Column(T.I32, lens, c.validity)


def _f_length(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if not isinstance(c, StringColumn):
        raise NotImplementedError("length on non-string")
    lens = _utf8_char_lens(c.offsets, c.bytes_)
    if c.is_dict:
        return Column(T.I32, lens[c.codes.long().clamp_min(0)], c.validity)
    return Column(T.I32, lens, c.validity)


def _utf8_char_lens(offsets: torch.Tensor, bytes_: torch.Tensor) -> torch.Tensor:
    """Character count per string: continuation bytes (0b10xxxxxx) don't
    start a character; prefix-sum the starts and difference at offsets."""
    if bytes_.numel() == 0:
        return torch.zeros(offsets.numel() - 1, dtype=torch.int32, device=offsets.device)
    starts = ((bytes_ & 0xC0) != 0x80).to(torch.int32)
    cum = torch.zeros(bytes_.numel() + 1, dtype=torch.int64, device=offsets.device)
    torch.cumsum(starts.to(torch.int64), 0, out=cum[1:])
    return (cum.index_select(0, offsets[1:]) - cum.index_select(0, offsets[:-1])).to(torch.int32)


def _f_substring(args, out, chunk, ev):
    c = _col(args[0], chunk)
    start = _scalarize(args[1])
    length = _scalarize(args[2]) if len(args) > 2 else None
    if start is None:
        raise NotImplementedError("substring with non-literal start")
    s = start.value
    ln = length.value if length is not None else None

    def sub(v: str) -> str:
        if s > 0:
            i = s - 1
        elif s == 0:
            i = 0
        else:
            i = max(len(v) + s, 0)
        return v[i : i + ln] if ln is not None else v[i:]

    if isinstance(c, StringColumn) and c.is_dict:
        vals = [sub(v) for v in c.dict_values()]
        from .column import _pack_strings

        offs, byts = _pack_strings(vals, c.device)
        return StringColumn(offs, byts, c.validity, c.codes)
    if isinstance(c, StringColumn) and c.is_cuda and s >= 1 and ln is not None:
        # device kernel: fixed-pitch extraction then compaction (all on GPU)
        from ..ops import kernels as K

        buf, lens = K.require().substr_fixed(c.offsets, c.bytes_, s - 1, ln)
        n = len(c)
        lens64 = lens.to(torch.int64)
        if bool((lens64 == ln).all()):
            offsets = torch.arange(0, (n + 1) * ln, ln, dtype=torch.int64, device=c.device)
            return StringColumn(offsets, buf, c.validity)
        mask = (torch.arange(ln, device=c.device).unsqueeze(0) < lens64.unsqueeze(1)).reshape(-1)
        out_bytes = buf[mask]
        offsets = torch.zeros(n + 1, dtype=torch.int64, device=c.device)
        torch.cumsum(lens64, 0, out=offsets[1:])
        return StringColumn(offsets, out_bytes, c.validity)
    return _str_map(c, sub)


def _f_concat(args, out, chunk, ev):
    from .eval import Scalar

    from .column import ListColumn

    if args and any(isinstance(a, ListColumn)
                    or isinstance(getattr(a, "dtype", None), T.ArrayType)
                    for a in args):
        return _IMPLS["array_concat"](args, out, chunk, ev)
    n = chunk.num_rows
    parts = []
    for a in args:
        if isinstance(a, Scalar):
            parts.append([a.value] * n)
        else:
            parts.append(_col(a, chunk).to_pylist())
    res = []
    for i in range(n):
        vs = [p[i] for p in parts]
        res.append(None if any(v is None for v in vs) else "".join(str(v) for v in vs))
    return StringColumn.from_pylist(res, device=chunk.device)


def _str_pred(fn):
    def impl(args, out, chunk, ev):
        c = _col(args[0], chunk)
        pat = _scalarize(args[1])
        if pat is None:
            raise NotImplementedError("string predicate with column pattern")
        p = pat.value
        if isinstance(c, StringColumn) and c.is_dict:
            if c.is_cuda and "%" not in p and "_" not in p:
                from ..ops import kernels as K

                like_pat = {"startswith": p + "%", "endswith": "%" + p,
                            "contains": "%" + p + "%"}[fn.__name__]
                hit = K.require().like_mask(c.offsets, c.bytes_, like_pat.encode())
            else:
                hit = torch.tensor([fn(v, p) for v in c.dict_values()],
                                   dtype=torch.bool, device=c.device)
            return Column(T.BOOL, hit[c.codes.long().clamp_min(0)] & (c.codes >= 0), c.validity)
        if c.is_cuda:
            from ..ops import kernels as K

            m = K.string_predicate(c, fn.__name__, p)
            if m is not None:
                return Column(T.BOOL, m, c.validity)
        vals = c.to_pylist()
        return Column(T.BOOL, torch.tensor([fn(v, p) if v is not None else False for v in vals],
                                           dtype=torch.bool, device=c.device), c.validity)

    return impl


def startswith(v, p):
    return v.startswith(p)


def endswith(v, p):
    return v.endswith(p)


def contains(v, p):
    return p in v


def _f_trim(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return _str_map(c, str.strip)


def _f_replace(args, out, chunk, ev):
    c = _col(args[0], chunk)
    old = _scalarize(args[1]).value
    new = _scalarize(args[2]).value if len(args) > 2 else ""
    return _str_map(c, lambda v: v.replace(old, new))


def _f_split_part(args, out, chunk, ev):
    c = _col(args[0], chunk)
    delim = _scalarize(args[1]).value
    part = _scalarize(args[2]).value

    def sp(v):
        parts = v.split(delim)
        i = part - 1 if part > 0 else len(parts) + part
        return parts[i] if 0 <= i < len(parts) else ""

    return _str_map(c, sp)


def _f_locate(args, out, chunk, ev):
    """locate/position(substr, str[, pos]) — substring FIRST (Spark)."""
    sub_s = _scalarize(args[0])
    c = _col(args[1], chunk)
    start = 1
    if len(args) > 2 and _scalarize(args[2]) is not None:
        start = int(_scalarize(args[2]).value)
    if sub_s is None:
        subs = _col(args[0], chunk).to_pylist()
        vals = c.to_pylist()
        res = [None if (v is None or s is None)
               else v.find(s, max(start - 1, 0)) + 1
               for v, s in zip(vals, subs)]
        return Column.from_values(res, T.I32, device=chunk.device)
    sub = sub_s.value
    res = _str_map(c, lambda v: v.find(sub, max(start - 1, 0)) + 1,
                   out_is_string=False)
    return Column.from_values(res, T.I32, device=chunk.device)


def _f_instr(args, out, chunk, ev):
    c = _col(args[0], chunk)
    sub = _scalarize(args[1]).value
    res = _str_map(c, lambda v: v.find(sub) + 1, out_is_string=False)
    return Column.from_values(res, T.I32, device=chunk.device)


# ---------------------------------------------------------------------------
# hashing / misc
# ---------------------------------------------------------------------------

def _f_xxhash64(args, out, chunk, ev):
    # engine-internal hash (shuffle partitioning); not Spark-bit-exact yet
    cols = [_col(a, chunk) for a in args]
    acc = torch.zeros(chunk.num_rows, dtype=torch.int64, device=chunk.device)
    for c in cols:
        if isinstance(c, StringColumn):
            data = c.codes.to(torch.int64) if c.is_dict else _cheap_string_hash(c)
        else:
            data = c.data.to(torch.int64) if c.data.dtype != torch.float64 else c.data.view(torch.int64)
        acc = acc * 31 + data
        mixed = acc ^ (acc >> 33)
        mixed = mixed * -49064778989728563  # 0xFF51AFD7ED558CCD as signed
        acc = mixed ^ (mixed >> 33)
    return Column(T.I64, acc, None)


def _cheap_string_hash(c: StringColumn) -> torch.Tensor:
    vals = c.to_pylist()
    return torch.tensor([hash(v) if v is not None else 0 for v in vals],
                        dtype=torch.int64, device=c.device)


def _f_monotonic_id(args, out, chunk, ev):
    return Column(T.I64, torch.arange(chunk.num_rows, dtype=torch.int64, device=chunk.device), None)


def _f_current_date(args, out, chunk, ev):
    from .eval import Scalar

    return Scalar((_dt.date.today() - _EPOCH).days, T.DATE)


def _f_current_timestamp(args, out, chunk, ev):
    from .eval import Scalar
    import time

    return Scalar(int(time.time() * 1e6), T.TIMESTAMP)


def _f_typeof(args, out, chunk, ev):
    from .eval import Scalar

    a = args[0]
    return Scalar(repr(a.dtype), T.STRING)


def _cast_fn(target):
    def impl(args, out, chunk, ev):
        from .eval import cast_value

        return cast_value(args[0], target, chunk)

    return impl


_IMPLS = {
    # datetime
    "year": _f_year, "month": _f_month, "day": _f_day, "dayofmonth": _f_day,
    "quarter": _f_quarter, "dayofweek": _f_dayofweek, "weekday": _f_weekday,
    "dayofyear": _f_dayofyear, "hour": _f_hour, "minute": _f_minute, "second": _f_second,
    "date_add": _f_date_add, "dateadd": _f_date_add, "date_sub": _f_date_sub,
    "datediff": _f_datediff, "date_diff": _f_datediff, "add_months": _f_add_months,
    "last_day": _f_last_day, "trunc": _f_trunc, "date_trunc": _f_date_trunc,
    "to_date": _f_to_date, "make_date": _f_make_date,
    "current_date": _f_current_date, "curdate": _f_current_date,
    "current_timestamp": _f_current_timestamp, "now": _f_current_timestamp,
    # math
    "abs": _f_abs, "round": _f_round, "floor": _f_floor, "ceil": _f_ceil,
    "ceiling": _f_ceil, "power": _f_power, "pow": _f_power,
    "greatest": _f_greatest, "least": _f_least, "sign": _f_sign, "signum": _f_sign,
    "isnan": _f_isnan,
    "sqrt": _unary_float(torch.sqrt), "exp": _unary_float(torch.exp),
    "ln": _unary_float(torch.log), "log": _unary_float(torch.log),
    "log10": _unary_float(torch.log10), "log2": _unary_float(torch.log2),
    "log1p": _unary_float(torch.log1p), "expm1": _unary_float(torch.expm1),
    "sin": _unary_float(torch.sin), "cos": _unary_float(torch.cos),
    "tan": _unary_float(torch.tan), "asin": _unary_float(torch.asin),
    "acos": _unary_float(torch.acos), "atan": _unary_float(torch.atan),
    "sinh": _unary_float(torch.sinh), "cosh": _unary_float(torch.cosh),
    "tanh": _unary_float(torch.tanh), "cbrt": _unary_float(lambda x: torch.sign(x) * torch.abs(x) ** (1 / 3)),
    "degrees": _unary_float(torch.rad2deg), "radians": _unary_float(torch.deg2rad),
    "rint": _unary_float(torch.round),
    # conditional
    "coalesce": _f_coalesce, "nvl": _f_coalesce, "ifnull": _f_coalesce,
    "if": _f_if, "iff": _f_if, "nullif": _f_nullif,
    # strings
    "upper": _f_upper, "ucase": _f_upper, "lower": _f_lower, "lcase": _f_lower,
    "length": _f_length, "len": _f_length, "char_length": _f_length,
    "character_length": _f_length, "octet_length": _f_octet_length,
    "substring": _f_substring, "substr": _f_substring, "concat": _f_concat,
    "startswith": _str_pred(startswith), "endswith": _str_pred(endswith),
    "contains": _str_pred(contains), "trim": _f_trim,
    "replace": _f_replace, "split_part": _f_split_part, "instr": _f_instr,
    "locate": _f_locate,
    # misc
    "xxhash64": _f_xxhash64, "hash": _f_xxhash64,
    "monotonically_increasing_id": _f_monotonic_id,
    "typeof": _f_typeof,
    # conversion helpers
    "double": _cast_fn(T.F64), "float": _cast_fn(T.F32), "int": _cast_fn(T.I32),
    "integer": _cast_fn(T.I32), "bigint": _cast_fn(T.I64), "long": _cast_fn(T.I64),
    "string": _cast_fn(T.STRING), "date": _cast_fn(T.DATE),
    "timestamp": _cast_fn(T.TIMESTAMP), "boolean": _cast_fn(T.BOOL),
}


def _f_regexp_replace(args, out, chunk, ev):
    import re as _re

    c = _col(args[0], chunk)
    pat = _scalarize(args[1]).value
    repl = _scalarize(args[2]).value
    # SQL backreferences \1 -> python \1 (compatible); $1 also accepted
    py_repl = repl.replace("$1", "\\1").replace("$2", "\\2")
    rx = _re.compile(pat)

    def f(v):
        return rx.sub(py_repl, v)

    if isinstance(c, StringColumn) and c.is_dict:
        vals = [f(v) for v in c.dict_values()]
        # transformed values may collide: re-dictionary (sorted) + remap codes
        uniq = sorted(set(vals))
        idx = {s: i for i, s in enumerate(uniq)}
        import torch as _t

        lut = _t.tensor([idx[v] for v in vals], dtype=_t.int32, device=c.device)
        from .column import _pack_strings

        offs, byts = _pack_strings(uniq, c.device)
        codes = lut[c.codes.to(_t.int64).clamp_min(0)]
        codes = _t.where(c.codes >= 0, codes, c.codes)
        return StringColumn(offs, byts, c.validity, codes)
    return _str_map(c, f)


def _f_length_any(args, out, chunk, ev):
    return _f_length(args, out, chunk, ev)


_IMPLS["regexp_replace"] = _f_regexp_replace


# ---------------------------------------------------------------------------
# breadth batch: string functions via per-dictionary host transforms (dict
# columns transform |dict| values, not |rows|) with raw-string fallback
# ---------------------------------------------------------------------------

def _dict_transform(fn):
    """Wrap a str->str transform as a column impl preserving dict encoding."""

    def impl(args, out, chunk, ev):
        c = _col(args[0], chunk)
        extra = [(_scalarize(a).value if _scalarize(a) else None) for a in args[1:]]

        def f(v):
            return fn(v, *extra)

        if isinstance(c, StringColumn) and c.is_dict:
            vals = [f(v) for v in c.dict_values()]
            uniq = sorted({v for v in vals if v is not None})
            idx = {s: i for i, s in enumerate(uniq)}
            # fn may return None for some dictionary entries: those become
            # code -1 + invalid rows
            lut = torch.tensor([-1 if v is None else idx[v] for v in vals],
                               dtype=torch.int32, device=c.device)
            from .column import _pack_strings

            offs, byts = _pack_strings(uniq, c.device)
            codes = lut[c.codes.to(torch.int64).clamp_min(0)]
            codes = torch.where(c.codes >= 0, codes, c.codes)
            validity = c.validity
            if bool((lut < 0).any()):
                valid = codes >= 0
                if validity is not None:
                    valid &= c.valid_mask()
                validity = valid.to(torch.uint8)
            return StringColumn(offs, byts, validity, codes)
        return _str_map(c, f)

    return impl


def _dict_to_int(fn, out_type=None):
    def impl(args, out, chunk, ev):
        c = _col(args[0], chunk)
        extra = [(_scalarize(a).value if _scalarize(a) else None) for a in args[1:]]

        def f(v):
            return fn(v, *extra)

        ot = out_type or out or T.I32
        if isinstance(c, StringColumn) and c.is_dict:
            vals = [f(v) for v in c.dict_values()]
            lut = torch.tensor(vals, dtype=ot.storage, device=c.device)
            return Column(ot, lut[c.codes.to(torch.int64).clamp_min(0)], c.validity)
        res = _str_map(c, f, out_is_string=False)
        return Column.from_values(res, ot, device=chunk.device)

    return impl


def _f_concat_ws(args, out, chunk, ev):
    sep = _scalarize(args[0]).value
    n = chunk.num_rows
    parts = [_col(a, chunk).to_pylist() for a in args[1:]]
    res = [sep.join(str(v) for v in row if v is not None) for row in zip(*parts)]
    return StringColumn.from_pylist(res, device=chunk.device)


def _f_regexp_extract(args, out, chunk, ev):
    import re as _re

    pat = _re.compile(_scalarize(args[1]).value)
    gi = _scalarize(args[2]).value if len(args) > 2 else 1

    def f(v):
        m = pat.search(v)
        if not m:
            return ""
        return m.group(gi) if gi <= (m.lastindex or 0) else ("" if gi else m.group(0))

    return _dict_transform(lambda v: f(v))(args[:1], out, chunk, ev)


def _f_md5(args, out, chunk, ev):
    import hashlib

    return _dict_transform(lambda v: hashlib.md5(v.encode()).hexdigest())(args[:1], out, chunk, ev)


def _f_sha2(args, out, chunk, ev):
    import hashlib

    bits = _scalarize(args[1]).value if len(args) > 1 else 256
    algo = {0: "sha256", 224: "sha224", 256: "sha256", 384: "sha384", 512: "sha512"}[bits]

    def f(v):
        return getattr(hashlib, algo)(v.encode()).hexdigest()

    return _dict_transform(lambda v: f(v))(args[:1], out, chunk, ev)


def _f_months_between(args, out, chunk, ev):
    a = _col(args[0], chunk)
    b = _col(args[1], chunk)
    ya, ma, da = _civil_from_days(_as_days(a))
    yb, mb, db = _civil_from_days(_as_days(b))
    months = (ya - yb) * 12 + (ma - mb)
    frac = (da - db).to(torch.float64) / 31.0
    return Column(T.F64, months.to(torch.float64) + frac, _merge(a, b))


def _f_next_day(args, out, chunk, ev):
    c = _col(args[0], chunk)
    dow_name = _scalarize(args[1]).value.lower()[:3]
    target = {"sun": 1, "mon": 2, "tue": 3, "wed": 4, "thu": 5, "fri": 6, "sat": 7}[dow_name]
    days = _as_days(c)
    cur = torch.remainder(days + 4, 7) + 1  # 1=Sunday
    delta = torch.remainder(target - cur + 7 - 1, 7) + 1
    return Column(T.DATE, (days + delta).to(torch.int32), c.validity)


def _f_weekofyear(args, out, chunk, ev):
    c = _col(args[0], chunk)
    days = _as_days(c)
    # ISO week number: Thursday-of-week determines the year
    dow = torch.remainder(days + 3, 7)  # 0=Monday
    thursday = days - dow + 3
    y, m, d = _civil_from_days(thursday)
    jan1 = _days_from_civil(y, torch.ones_like(y), torch.ones_like(y))
    week = torch.div(thursday - jan1, 7, rounding_mode="floor") + 1
    return Column(T.I32, week.to(torch.int32), c.validity)


def _f_unix_timestamp(args, out, chunk, ev):
    if not args:
        import time as _time

        from .eval import Scalar

        return Scalar(int(_time.time()), T.I64)
    c = _col(args[0], chunk)
    if isinstance(c.dtype, T.TimestampType):
        return Column(T.I64, torch.div(c.data, 1_000_000, rounding_mode="floor"), c.validity)
    if isinstance(c.dtype, T.DateType):
        return Column(T.I64, c.data.to(torch.int64) * 86400, c.validity)
    # strings: parse (optional Spark pattern) then convert to seconds
    ts = _f_to_timestamp(args, T.TIMESTAMP, chunk, ev)
    return Column(T.I64, torch.div(ts.data, 1_000_000, rounding_mode="floor"),
                  ts.validity)


def _f_from_unixtime_ts(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return Column(T.TIMESTAMP, c.data.to(torch.int64) * 1_000_000, c.validity)


def _f_make_timestamp(args, out, chunk, ev):
    y = _col(args[0], chunk).data.to(torch.int64)
    mo = _col(args[1], chunk).data.to(torch.int64)
    d = _col(args[2], chunk).data.to(torch.int64)
    h = _col(args[3], chunk).data.to(torch.int64)
    mi = _col(args[4], chunk).data.to(torch.int64)
    se = _col(args[5], chunk).data.to(torch.float64)
    days = _days_from_civil(y, mo, d)
    us = (days * 86400 + h * 3600 + mi * 60) * 1_000_000 + (se * 1e6).to(torch.int64)
    return Column(T.TIMESTAMP, us, None)


def _f_nvl2(args, out, chunk, ev):
    from .eval import cast_value

    a = _col(args[0], chunk)
    when_ok = _col(cast_value(args[1], out, chunk), chunk)
    when_null = _col(cast_value(args[2], out, chunk), chunk)
    ok = a.valid_mask()
    data = torch.where(ok, when_ok.data, when_null.data.to(when_ok.data.dtype))
    valid = torch.where(ok, when_ok.valid_mask(), when_null.valid_mask())
    return Column(out, data, None if bool(valid.all()) else valid.to(torch.uint8))


def _f_bit_ops(op):
    def impl(args, out, chunk, ev):
        a = _col(args[0], chunk)
        if len(args) == 1:
            return Column(a.dtype, ~a.data, a.validity)
        b = _col(args[1], chunk)
        x, y = a.data.to(torch.int64), b.data.to(torch.int64)
        data = {"shl": x << y, "shr": x >> y, "and": x & y, "or": x | y, "xor": x ^ y}[op]
        return Column(T.I64, data, _merge(a, b))

    return impl


def _f_hex(args, out, chunk, ev):
    c = _col(args[0], chunk)
    if isinstance(c, StringColumn):
        return _dict_transform(
            lambda v: (v if isinstance(v, (bytes, bytearray))
                       else v.encode()).hex().upper())(args, out, chunk, ev)
    vals = c.to_pylist()
    return StringColumn.from_pylist([format(int(v), "X") if v is not None else None for v in vals],
                                    device=chunk.device)


def _f_factorial(args, out, chunk, ev):
    import math as _m

    c = _col(args[0], chunk)
    lut = torch.tensor([_m.factorial(i) for i in range(21)], dtype=torch.int64, device=c.device)
    x = c.data.to(torch.int64).clamp(0, 20)
    return Column(T.I64, lut[x], c.validity)


def _f_pi(args, out, chunk, ev):
    from .eval import Scalar

    return Scalar(math.pi, T.F64)


def _f_e(args, out, chunk, ev):
    from .eval import Scalar

    return Scalar(math.e, T.F64)


def _f_format_number(args, out, chunk, ev):
    c = _col(args[0], chunk)
    nd = _scalarize(args[1]).value
    vals = c.to_pylist()
    return StringColumn.from_pylist(
        [None if v is None else f"{v:,.{nd}f}" for v in vals], device=chunk.device)


def _f_atan2(args, out, chunk, ev):
    a = _col(args[0], chunk)
    b = _col(args[1], chunk)
    return Column(T.F64, torch.atan2(a.data.to(torch.float64), b.data.to(torch.float64)),
                  _merge(a, b))


def _f_log_base(args, out, chunk, ev):
    if len(args) == 1:
        c = _col(args[0], chunk)
        return Column(T.F64, torch.log(c.data.to(torch.float64)), c.validity)
    base = _scalarize(args[0]).value
    c = _col(args[1], chunk)
    return Column(T.F64, torch.log(c.data.to(torch.float64)) / math.log(base), c.validity)


_IMPLS.update({
    "ltrim": _dict_transform(lambda v, chars=None: v.lstrip(chars)),
    "rtrim": _dict_transform(lambda v, chars=None: v.rstrip(chars)),
    "btrim": _dict_transform(lambda v, chars=None: v.strip(chars)),
    "reverse": _dict_transform(lambda v: v[::-1]),
    "repeat": _dict_transform(lambda v, n: v * int(n)),
    "initcap": _dict_transform(lambda v: " ".join(w.capitalize() for w in v.split(" "))),
    "lpad": _dict_transform(lambda v, n, p=" ": v[:int(n)] if len(v) >= int(n) else (p * int(n))[: int(n) - len(v)] + v),
    "rpad": _dict_transform(lambda v, n, p=" ": v[:int(n)] if len(v) >= int(n) else v + (p * int(n))[: int(n) - len(v)]),
    "left": _dict_transform(lambda v, n: v[:int(n)]),
    "right": _dict_transform(lambda v, n: v[-int(n):] if int(n) else ""),
    "translate": _dict_transform(lambda v, frm, to: v.translate(str.maketrans(frm[:len(to)], to[:len(frm)]))),
    "substring_index": _dict_transform(
        lambda v, d, n: d.join(v.split(d)[:int(n)]) if int(n) > 0 else d.join(v.split(d)[int(n):])),
    "soundex": _dict_transform(lambda v: _soundex(v)),
    "ascii": _dict_to_int(lambda v: ord(v[0]) if v else 0),
    "position": _f_locate,
    "levenshtein": _dict_to_int(lambda v, w: _levenshtein(v, w)),
    "concat_ws": _f_concat_ws,
    "regexp_extract": _f_regexp_extract,
    "md5": _f_md5,
    "sha2": _f_sha2,
    "chr": _dict_transform(lambda v: v),  # placeholder; numeric chr below
    "months_between": _f_months_between,
    "next_day": _f_next_day,
    "weekofyear": _f_weekofyear,
    "unix_timestamp": _f_unix_timestamp, "to_unix_timestamp": _f_unix_timestamp,
    "timestamp_seconds": _f_from_unixtime_ts,
    "make_timestamp": _f_make_timestamp,
    "nvl2": _f_nvl2,
    "shiftleft": _f_bit_ops("shl"), "shiftright": _f_bit_ops("shr"),
    "bitwise_not": _f_bit_ops("not"),
    "hex": _f_hex,
    "factorial": _f_factorial,
    "pi": _f_pi, "e": _f_e,
    "format_number": _f_format_number,
    "atan2": _f_atan2,
    "log": _f_log_base,
})


def _f_chr(args, out, chunk, ev):
    c = _col(args[0], chunk)
    vals = c.to_pylist()
    return StringColumn.from_pylist(
        [None if v is None else chr(int(v) % 256) for v in vals], device=chunk.device)


_IMPLS["char"] = _f_chr
_IMPLS["chr"] = _f_chr


def _soundex(v: str) -> str:
    if not v:
        return ""
    codes = {"b": "1", "f": "1", "p": "1", "v": "1", "c": "2", "g": "2", "j": "2",
             "k": "2", "q": "2", "s": "2", "x": "2", "z": "2", "d": "3", "t": "3",
             "l": "4", "m": "5", "n": "5", "r": "6"}
    s = v.lower()
    out = v[0].upper()
    prev = codes.get(s[0], "")
    for ch in s[1:]:
        code = codes.get(ch, "")
        if code and code != prev:
            out += code
            if len(out) == 4:
                break
        prev = code if ch not in "hw" else prev
    return (out + "000")[:4]


def _levenshtein(a: str, b: str) -> int:
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + (ca != cb)))
        prev = cur
    return prev[-1]


# array/list functions (segment ops over ListColumn; engine/arrays.py)
from .arrays import IMPLS as _ARRAY_IMPLS  # noqa: E402

_IMPLS.update(_ARRAY_IMPLS)
_IMPLS["element_at"] = _ARRAY_IMPLS["element_at"]
_IMPLS["element_at_sql"] = _ARRAY_IMPLS["element_at_sql"]
_IMPLS["try_element_at"] = _ARRAY_IMPLS["try_element_at"]


# -- JSON functions (host eval, dictionary-aware — ref: sail-function
#    src/scalar/json) ---------------------------------------------------

def _json_path_get(doc: str, path: str):
    import json as _json

    if doc is None:
        return None
    try:
        obj = _json.loads(doc)
    except (ValueError, TypeError):
        return None
    if not path.startswith("$"):
        return None
    # $.a.b[0].c subset of JSONPath
    import re as _re

    for part in _re.findall(r"\.([A-Za-z_][A-Za-z0-9_]*)|\[(\d+)\]", path[1:]):
        key, idx = part
        if key:
            if not isinstance(obj, dict) or key not in obj:
                return None
            obj = obj[key]
        else:
            i = int(idx)
            if not isinstance(obj, list) or i >= len(obj):
                return None
            obj = obj[i]
    if obj is None:
        return None
    if isinstance(obj, (dict, list)):
        import json as _json

        return _json.dumps(obj, separators=(",", ":"))
    if isinstance(obj, bool):
        return "true" if obj else "false"
    return str(obj)


def _f_get_json_object(args, out, chunk, ev):
    path = _scalarize(args[1]).value
    return _dict_transform(lambda v: _json_path_get(v, path))(args[:1], out, chunk, ev)


def _f_json_tuple_field(args, out, chunk, ev):
    # json_tuple is exposed as get_json_object('$.<field>') per field
    field = _scalarize(args[1]).value
    return _dict_transform(lambda v: _json_path_get(v, f"$.{field}"))(args[:1], out, chunk, ev)


def _f_to_json(args, out, chunk, ev):
    import json as _json

    c = _col(args[0], chunk)
    vals = c.to_pylist()
    return StringColumn.from_pylist(
        [None if v is None else _json.dumps(v, separators=(",", ":"), default=str)
         for v in vals], device=chunk.device)


def _f_schema_of_json(args, out, chunk, ev):
    import json as _json

    from .eval import Scalar

    doc = _scalarize(args[0]).value

    def tname(v):
        if isinstance(v, bool):
            return "BOOLEAN"
        if isinstance(v, int):
            return "BIGINT"
        if isinstance(v, float):
            return "DOUBLE"
        if isinstance(v, list):
            return f"ARRAY<{tname(v[0]) if v else 'STRING'}>"
        if isinstance(v, dict):
            inner = ", ".join(f"{k}: {tname(x)}" for k, x in v.items())
            return f"STRUCT<{inner}>"
        return "STRING"

    try:
        return Scalar(tname(_json.loads(doc)), T.STRING)
    except (ValueError, TypeError):
        return Scalar(None, T.STRING)


_IMPLS["get_json_object"] = _f_get_json_object
_IMPLS["to_json"] = _f_to_json
_IMPLS["schema_of_json"] = _f_schema_of_json


def _f_from_json(args, out, chunk, ev):
    """from_json(col, 'a INT, b STRING') -> struct (host JSON parse)."""
    import json as _json

    from .column import StructColumn

    c = _col(args[0], chunk)
    docs = c.to_pylist()
    parsed = []
    for d in docs:
        if d is None:
            parsed.append(None)
            continue
        try:
            v = _json.loads(d)
            parsed.append(v if isinstance(v, dict) else None)
        except (ValueError, TypeError):
            parsed.append(None)
    kids = []
    for f in out.fields:
        vals = [None if p is None else p.get(f.name) for p in parsed]
        # stringify nested values for STRING fields
        if isinstance(f.dtype, T.StringType):
            vals = [None if v is None else
                    (v if isinstance(v, str) else _json.dumps(v)) for v in vals]
        kids.append((f.name, Column.from_values(vals, f.dtype, device=chunk.device)))
    validity = None
    if any(p is None for p in parsed):
        validity = torch.tensor([0 if p is None else 1 for p in parsed],
                                dtype=torch.uint8, device=chunk.device)
    return StructColumn(kids, validity, dtype=out)


_IMPLS["from_json"] = _f_from_json


# -- datetime formatting (host path; Spark pattern subset) --------------

def _spark_fmt_to_strftime(p: str) -> str:
    # longest-first to keep yyyy from matching yy twice
    subs = [("yyyy", "%Y"), ("yy", "%y"), ("MMMM", "%B"), ("MMM", "%b"),
            ("MM", "%m"), ("dd", "%d"), ("HH", "%H"), ("hh", "%I"),
            ("mm", "%M"), ("ss", "%S"), ("EEEE", "%A"), ("EEE", "%a"),
            ("a", "%p"), ("DDD", "%j")]
    out = ""
    i = 0
    while i < len(p):
        for pat, rep in subs:
            if p.startswith(pat, i):
                out += rep
                i += len(pat)
                break
        else:
            out += p[i]
            i += 1
    return out


def _f_date_format(args, out, chunk, ev):
    c = _col(args[0], chunk)
    fmt = _spark_fmt_to_strftime(_scalarize(args[1]).value)
    vals = c.to_pylist()
    if isinstance(c.dtype, T.TimestampType):
        # timestamps collect as epoch micros; format needs datetimes
        vals = [None if v is None else
                _dt.datetime(1970, 1, 1) + _dt.timedelta(microseconds=int(v))
                for v in vals]
    res = [None if v is None else v.strftime(fmt) if hasattr(v, "strftime")
           else str(v) for v in vals]
    return StringColumn.from_pylist(res, device=chunk.device)


def _f_from_unixtime(args, out, chunk, ev):
    c = _col(args[0], chunk)
    fmt = _spark_fmt_to_strftime(_scalarize(args[1]).value) \
        if len(args) > 1 and _scalarize(args[1]) else "%Y-%m-%d %H:%M:%S"
    vals = c.to_pylist()
    res = [None if v is None else
           _dt.datetime.fromtimestamp(int(v), _dt.timezone.utc).strftime(fmt)
           for v in vals]
    return StringColumn.from_pylist(res, device=chunk.device)


def _f_to_timestamp(args, out, chunk, ev):
    c = _col(args[0], chunk)
    fmt = _spark_fmt_to_strftime(_scalarize(args[1]).value) \
        if len(args) > 1 and _scalarize(args[1]) else None
    vals = c.to_pylist()
    res = []
    for v in vals:
        if v is None:
            res.append(None)
            continue
        try:
            if fmt:
                dt = _dt.datetime.strptime(v, fmt)
            else:
                dt = _dt.datetime.fromisoformat(str(v))
            res.append(int(dt.replace(tzinfo=_dt.timezone.utc).timestamp() * 1_000_000))
        except (ValueError, TypeError):
            res.append(None)
    validity = None
    if any(r is None for r in res):
        validity = torch.tensor([0 if r is None else 1 for r in res],
                                dtype=torch.uint8, device=chunk.device)
    data = torch.tensor([0 if r is None else r for r in res],
                        dtype=torch.int64, device=chunk.device)
    return Column(T.TIMESTAMP, data, validity)


def _f_datepart(args, out, chunk, ev):
    unit = str(_scalarize(args[0]).value).lower()
    fmap = {"year": "year", "yyyy": "year", "yy": "year", "month": "month",
            "mon": "month", "mm": "month", "day": "day", "dd": "day",
            "dayofweek": "dayofweek", "dow": "dayofweek", "doy": "dayofyear",
            "hour": "hour", "minute": "minute", "second": "second",
            "quarter": "quarter", "week": "weekofyear"}
    name = fmap.get(unit)
    if name is None:
        raise NotImplementedError(f"date_part unit {unit}")
    return dispatch_function(name, [args[1]], out, chunk, ev)


_IMPLS["date_format"] = _f_date_format
_IMPLS["from_unixtime"] = _f_from_unixtime
_IMPLS["to_timestamp"] = _f_to_timestamp
_IMPLS["try_to_timestamp"] = _f_to_timestamp
_IMPLS["datepart"] = _f_datepart
_IMPLS["date_part"] = _f_datepart



def _parse_duration_us(text: str) -> int:
    import re as _re

    m = _re.match(r"\s*(\d+)\s*(\w+)\s*$", str(text))
    if not m:
        raise ValueError(f"bad duration {text!r}")
    n, unit = int(m.group(1)), m.group(2).lower().rstrip("s")
    mult = {"microsecond": 1, "millisecond": 1000, "second": 1_000_000,
            "minute": 60_000_000, "hour": 3_600_000_000,
            "day": 86_400_000_000, "week": 7 * 86_400_000_000}
    if unit not in mult:
        raise ValueError(f"bad duration unit {unit!r}")
    return n * mult[unit]


def _f_window(args, out, chunk, ev):
    """Tumbling (or sliding start-aligned) event-time window
    (ref: Spark window() grouping function). Returns struct(start, end)
    in TIMESTAMP micros; GROUP BY window(ts, '1 hour') works because
    struct grouping falls back to its fields."""
    from .column import StructColumn

    c = _col(args[0], chunk)
    if len(args) > 2:
        slide = _parse_duration_us(_scalarize(args[2]).value)
        width = _parse_duration_us(_scalarize(args[1]).value)
        if slide != width:
            raise NotImplementedError(
                "sliding windows (slide != width) are not supported yet; "
                "use tumbling windows")
    width = _parse_duration_us(_scalarize(args[1]).value)
    us = c.data.to(torch.int64)
    if isinstance(c.dtype, T.DateType):
        us = us * 86_400_000_000
    start = torch.div(us, width, rounding_mode="floor") * width
    end = start + width
    return StructColumn([("start", Column(T.TIMESTAMP, start, c.validity)),
                         ("end", Column(T.TIMESTAMP, end, c.validity))],
                        c.validity, dtype=out)


def _f_window_time(args, out, chunk, ev):
    """window_time(w) = w.end - 1 microsecond (Spark semantics)."""
    c = _col(args[0], chunk)
    end = dict(c.children_)["end"]
    return Column(T.TIMESTAMP, end.data.to(torch.int64) - 1, c.validity)


_IMPLS["window"] = _f_window
_IMPLS["window_time"] = _f_window_time


# -- URL / XML xpath / CSV / variant families (host eval — ref:
#    sail-function src/scalar/{url,xml,csv,variant}) -----------------------

def _parse_url_part(url, part, key=None):
    from urllib.parse import urlparse, parse_qs

    if url is None or part is None:
        return None
    try:
        u = urlparse(url)
    except ValueError:
        return None
    part = part.upper()
    if part == "HOST":
        return u.hostname
    if part == "PATH":
        return u.path or None
    if part == "QUERY":
        if key is not None:
            vals = parse_qs(u.query, keep_blank_values=False).get(key)
            return vals[0] if vals else None
        return u.query or None
    if part == "REF":
        return u.fragment or None
    if part == "PROTOCOL":
        return u.scheme or None
    if part == "FILE":
        return (u.path + ("?" + u.query if u.query else "")) or None
    if part == "AUTHORITY":
        return u.netloc or None
    if part == "USERINFO":
        if "@" not in u.netloc:
            return None
        return u.netloc.rsplit("@", 1)[0]
    return None


def _f_parse_url(args, out, chunk, ev):
    part = _scalarize(args[1]).value
    key = _scalarize(args[2]).value if len(args) > 2 else None
    return _dict_transform(
        lambda v: _parse_url_part(v, part, key))(args[:1], out, chunk, ev)


def _f_url_encode(args, out, chunk, ev):
    from urllib.parse import quote_plus

    return _dict_transform(
        lambda v: None if v is None else quote_plus(v))(args[:1], out, chunk, ev)


def _f_url_decode(args, out, chunk, ev):
    from urllib.parse import unquote_plus

    return _dict_transform(
        lambda v: None if v is None else unquote_plus(v))(args[:1], out, chunk, ev)


def _xpath_nodes(doc, path):
    """Subset of XPath over ElementTree: steps a/b, .//b, [@k='v'] filters,
    trailing /text() or /@attr."""
    import xml.etree.ElementTree as ET

    if doc is None:
        return None
    want_text = False
    attr = None
    if path.endswith("/text()"):
        want_text, path = True, path[: -len("/text()")]
    elif "/@" in path:
        path, attr = path.rsplit("/@", 1)
    try:
        root = ET.fromstring(doc)
    except ET.ParseError:
        return None
    # XPath evaluates from the DOCUMENT node (Java semantics): both '/a/b'
    # and 'a/b' take their first step against the root element's tag
    if path.startswith("//"):
        path = "." + path  # '//c' -> './/c' (descendant search)
    elif path:
        steps = path.lstrip("/").split("/", 1)
        if steps[0] not in (root.tag, "*", "."):
            return []
        path = steps[1] if len(steps) > 1 else "."
    nodes = root.findall(path) if path else [root]
    if attr is not None:
        return [n.get(attr) for n in nodes if n.get(attr) is not None]
    return [(n.text or "") for n in nodes] if (want_text or True) else nodes


def _xpath_first(doc, path):
    got = _xpath_nodes(doc, path)
    return got[0] if got else None


def _f_xpath(args, out, chunk, ev):
    from .column import ListColumn, StringColumn

    path = _scalarize(args[1]).value
    c = _col(args[0], chunk)
    rows = [_xpath_nodes(v, path) for v in c.to_pylist()]
    flat = [x for r in rows if r for x in r]
    offs = [0]
    for r in rows:
        offs.append(offs[-1] + (len(r) if r else 0))
    validity = None
    if any(r is None for r in rows):
        validity = torch.tensor([0 if r is None else 1 for r in rows],
                                dtype=torch.uint8, device=chunk.device)
    child = StringColumn.from_pylist(flat, device=str(chunk.device))
    return ListColumn(torch.tensor(offs, dtype=torch.int64,
                                   device=chunk.device), child, validity, out)


def _f_xpath_typed(cast):
    def run(args, out, chunk, ev):
        path = _scalarize(args[1]).value
        c = _col(args[0], chunk)
        vals = [cast(_xpath_first(v, path)) for v in c.to_pylist()]
        return Column.from_values(vals, out, device=str(chunk.device))
    return run


def _xp_num(v):
    if v is None or v == "":
        return None
    try:
        return float(v)
    except ValueError:
        return None


def _xp_int(v):
    f = _xp_num(v)
    return None if f is None else int(f)


def _f_xpath_boolean(args, out, chunk, ev):
    path = _scalarize(args[1]).value
    c = _col(args[0], chunk)
    vals = [None if (r := _xpath_nodes(v, path)) is None else bool(r)
            for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


def _csv_split(line, sep):
    import csv as _csv
    import io as _io

    if line is None:
        return None
    return next(_csv.reader(_io.StringIO(line), delimiter=sep))


def _f_from_csv(args, out, chunk, ev):
    from .column import StructColumn

    sep = ","
    if len(args) > 2:
        opts = _scalarize(args[2]).value
        if isinstance(opts, str) and opts:
            import json as _json

            try:
                sep = _json.loads(opts).get("sep", ",")
            except ValueError:
                pass
    c = _col(args[0], chunk)
    rows = [_csv_split(v, sep) for v in c.to_pylist()]
    kids = []
    for i, f in enumerate(out.fields):
        raw = [None if r is None or i >= len(r) or r[i] == "" else r[i]
               for r in rows]
        if not isinstance(f.dtype, T.StringType):
            conv = []
            for v in raw:
                if v is None:
                    conv.append(None)
                else:
                    try:
                        conv.append(float(v) if isinstance(
                            f.dtype, (T.Float32Type, T.Float64Type,
                                      T.DecimalType)) else int(v))
                    except ValueError:
                        conv.append(None)
            raw = conv
        kids.append((f.name, Column.from_values(raw, f.dtype,
                                                device=str(chunk.device))))
    validity = None
    if any(r is None for r in rows):
        validity = torch.tensor([0 if r is None else 1 for r in rows],
                                dtype=torch.uint8, device=chunk.device)
    return StructColumn(kids, validity, dtype=out)


def _f_to_csv(args, out, chunk, ev):
    from .column import StringColumn

    c = _col(args[0], chunk)
    names = [nm for nm, _ in c.children_]
    cols = [kid.to_pylist() for _, kid in c.children_]
    n = len(c)
    vals = []
    for i in range(n):
        parts = []
        for j in range(len(names)):
            v = cols[j][i]
            if v is None:
                parts.append("")
            elif isinstance(v, bool):
                parts.append("true" if v else "false")
            else:
                parts.append(str(v))
        vals.append(",".join(parts))
    vcol = StringColumn.from_pylist(vals, device=str(chunk.device))
    vcol.validity = c.validity
    return vcol


def _f_schema_of_csv(args, out, chunk, ev):
    sample = _scalarize(args[0]).value
    parts = _csv_split(sample, ",") or []
    kinds = []
    for p in parts:
        try:
            int(p)
            kinds.append("BIGINT")
            continue
        except ValueError:
            pass
        try:
            float(p)
            kinds.append("DOUBLE")
            continue
        except ValueError:
            kinds.append("STRING")
    body = ", ".join(f"_c{i}: {k}" for i, k in enumerate(kinds))
    from .column import StringColumn

    return StringColumn.from_pylist([f"STRUCT<{body}>"] * chunk.num_rows,
                                    device=str(chunk.device))


def _canon_json(v, strict):
    import json as _json

    if v is None:
        return None
    try:
        return _json.dumps(_json.loads(v), separators=(",", ":"))
    except (ValueError, TypeError):
        if strict:
            raise ValueError(f"parse_json: malformed JSON: {v!r}")
        return None


def _f_parse_json(args, out, chunk, ev):
    return _dict_transform(lambda v: _canon_json(v, True))(args[:1], out, chunk, ev)


def _f_try_parse_json(args, out, chunk, ev):
    return _dict_transform(lambda v: _canon_json(v, False))(args[:1], out, chunk, ev)


def _f_variant_get(args, out, chunk, ev):
    # variant is stored as canonical JSON text (documented simplification);
    # a literal third `type` arg casts the result (Spark semantics)
    path = _scalarize(args[1]).value
    col = _dict_transform(
        lambda v: _json_path_get(v, path))(args[:1], out, chunk, ev)
    tn = _scalarize(args[2]) if len(args) > 2 else None
    if tn is not None and tn.value:
        from .eval import cast_column

        col = cast_column(col, T.type_from_name(str(tn.value)))
    return col


def _f_is_variant_null(args, out, chunk, ev):
    c = _col(args[0], chunk)
    vals = [None if v is None else (v.strip() == "null")
            for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


def _schema_of_variant_one(v):
    import json as _json

    if v is None:
        return None
    try:
        obj = _json.loads(v)
    except (ValueError, TypeError):
        return None

    def name(o):
        if o is None:
            return "VOID"
        if isinstance(o, bool):
            return "BOOLEAN"
        if isinstance(o, int):
            return "BIGINT"
        if isinstance(o, float):
            return "DOUBLE"
        if isinstance(o, str):
            return "STRING"
        if isinstance(o, list):
            inner = {name(x) for x in o} or {"VOID"}
            return f"ARRAY<{inner.pop() if len(inner) == 1 else 'VARIANT'}>"
        return ("OBJECT<" + ", ".join(
            f"{k}: {name(x)}" for k, x in sorted(o.items())) + ">")

    return name(obj)


def _f_schema_of_variant(args, out, chunk, ev):
    return _dict_transform(_schema_of_variant_one)(args[:1], out, chunk, ev)


def _f_json_array_length(args, out, chunk, ev):
    import json as _json

    def ln(v):
        if v is None:
            return None
        try:
            obj = _json.loads(v)
        except (ValueError, TypeError):
            return None
        return len(obj) if isinstance(obj, list) else None

    c = _col(args[0], chunk)
    vals = [ln(v) for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


def _f_json_object_keys(args, out, chunk, ev):
    import json as _json

    from .column import ListColumn, StringColumn

    c = _col(args[0], chunk)
    rows = []
    for v in c.to_pylist():
        try:
            obj = None if v is None else _json.loads(v)
        except (ValueError, TypeError):
            obj = None
        rows.append(list(obj.keys()) if isinstance(obj, dict) else None)
    flat = [k for r in rows if r for k in r]
    offs = [0]
    for r in rows:
        offs.append(offs[-1] + (len(r) if r else 0))
    validity = None
    if any(r is None for r in rows):
        validity = torch.tensor([0 if r is None else 1 for r in rows],
                                dtype=torch.uint8, device=chunk.device)
    return ListColumn(torch.tensor(offs, dtype=torch.int64, device=chunk.device),
                      StringColumn.from_pylist(flat, device=str(chunk.device)),
                      validity, out)


def _luhn_ok(v):
    if v is None or not v.isdigit() or not v:
        return False if v is not None else None
    total = 0
    for i, ch in enumerate(reversed(v)):
        d = ord(ch) - 48
        if i % 2 == 1:
            d *= 2
            if d > 9:
                d -= 9
        total += d
    return total % 10 == 0


def _f_luhn_check(args, out, chunk, ev):
    c = _col(args[0], chunk)
    vals = [_luhn_ok(v) for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


_CRC32C_TABLE = None


def _crc32c(data: bytes) -> int:
    global _CRC32C_TABLE
    if _CRC32C_TABLE is None:
        tbl = []
        for i in range(256):
            c = i
            for _ in range(8):
                c = (c >> 1) ^ 0x82F63B78 if c & 1 else c >> 1
            tbl.append(c)
        _CRC32C_TABLE = tbl
    crc = 0xFFFFFFFF
    for b in data:
        crc = (crc >> 8) ^ _CRC32C_TABLE[(crc ^ b) & 0xFF]
    return crc ^ 0xFFFFFFFF


def _f_crc32c(args, out, chunk, ev):
    c = _col(args[0], chunk)
    vals = [None if v is None else _crc32c(v.encode()) for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


_IMPLS["parse_url"] = _f_parse_url
_IMPLS["try_parse_url"] = _f_parse_url
_IMPLS["url_encode"] = _f_url_encode
_IMPLS["url_decode"] = _f_url_decode
_IMPLS["xpath"] = _f_xpath
_IMPLS["xpath_string"] = _f_xpath_typed(lambda v: v)
_IMPLS["xpath_int"] = _f_xpath_typed(_xp_int)
_IMPLS["xpath_short"] = _f_xpath_typed(_xp_int)
_IMPLS["xpath_long"] = _f_xpath_typed(_xp_int)
_IMPLS["xpath_double"] = _f_xpath_typed(_xp_num)
_IMPLS["xpath_float"] = _f_xpath_typed(_xp_num)
_IMPLS["xpath_number"] = _f_xpath_typed(_xp_num)
_IMPLS["xpath_boolean"] = _f_xpath_boolean
_IMPLS["from_csv"] = _f_from_csv
_IMPLS["to_csv"] = _f_to_csv
_IMPLS["schema_of_csv"] = _f_schema_of_csv
_IMPLS["parse_json"] = _f_parse_json
_IMPLS["try_parse_json"] = _f_try_parse_json
_IMPLS["variant_get"] = _f_variant_get
_IMPLS["try_variant_get"] = _f_variant_get
_IMPLS["is_variant_null"] = _f_is_variant_null
_IMPLS["schema_of_variant"] = _f_schema_of_variant
_IMPLS["json_array_length"] = _f_json_array_length
_IMPLS["json_object_keys"] = _f_json_object_keys
_IMPLS["luhn_check"] = _f_luhn_check
_IMPLS["crc32c"] = _f_crc32c



def _f_array_concat(args, out, chunk, ev):
    from .arrays import _bcast, _concat_rows

    acc = _bcast(args[0], chunk)
    for a in args[1:]:
        acc = _concat_rows(acc, _bcast(a, chunk))
    return acc


_IMPLS["array_concat"] = _f_array_concat


# -- misc Spark scalar surface: regexp_*, mask/quote, timestamp arithmetic,
#    randoms, to_number/to_binary, bit access (ref: sail-function
#    src/scalar/{string,math,datetime,misc}) --------------------------------

def _f_strpos(args, out, chunk, ev):
    return _IMPLS["instr"](args, out, chunk, ev)


def _f_quote(args, out, chunk, ev):
    return _dict_transform(
        lambda v: None if v is None else
        "'" + v.replace("\\", "\\\\").replace("'", "\\'") + "'"
    )(args[:1], out, chunk, ev)


def _mask_one(v, up, lo, dig, other):
    if v is None:
        return None
    out = []
    for ch in v:
        if ch.isupper():
            out.append(ch if up is None else up)
        elif ch.islower():
            out.append(ch if lo is None else lo)
        elif ch.isdigit():
            out.append(ch if dig is None else dig)
        else:
            out.append(ch if other is None else other)
    return "".join(out)


def _f_mask(args, out, chunk, ev):
    def opt(i, default):
        if len(args) <= i:
            return default
        v = _scalarize(args[i]).value
        return v
    up = opt(1, "X")
    lo = opt(2, "x")
    dig = opt(3, "n")
    other = opt(4, None)
    return _dict_transform(
        lambda v: _mask_one(v, up, lo, dig, other))(args[:1], out, chunk, ev)


def _f_regexp_count(args, out, chunk, ev):
    import re as _re

    pat = _re.compile(_scalarize(args[1]).value)
    c = _col(args[0], chunk)
    vals = [None if v is None else len(pat.findall(v)) for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


def _f_regexp_instr(args, out, chunk, ev):
    import re as _re

    pat = _re.compile(_scalarize(args[1]).value)
    idx = int(_scalarize(args[2]).value) if len(args) > 2 else 0
    c = _col(args[0], chunk)
    vals = []
    for v in c.to_pylist():
        if v is None:
            vals.append(None)
            continue
        m = pat.search(v)
        vals.append(0 if m is None else
                    (m.start() + 1 if idx == 0 else m.end() + 1))
    return Column.from_values(vals, out, device=str(chunk.device))


def _f_regexp_substr(args, out, chunk, ev):
    import re as _re

    pat = _re.compile(_scalarize(args[1]).value)
    return _dict_transform(
        lambda v: None if v is None else
        (lambda m: m.group(0) if m else None)(pat.search(v))
    )(args[:1], out, chunk, ev)


_DAYNAMES = ["Mon", "Tue", "Wed", "Thu", "Fri", "Sat", "Sun"]


def _f_dayname(args, out, chunk, ev):
    from .column import StringColumn

    c = _col(args[0], chunk)
    days = c.data.to(torch.int64)
    if isinstance(c.dtype, T.TimestampType):
        days = torch.div(days, 86_400_000_000, rounding_mode="floor")
    dow = ((days % 7) + 3) % 7  # 1970-01-01 was a Thursday
    vals = [_DAYNAMES[int(d)] for d in dow.tolist()]
    col = StringColumn.from_pylist(vals, device=str(chunk.device))
    col.validity = c.validity
    return col


def _f_date_from_unix_date(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return Column(T.DATE, c.data.to(torch.int32), c.validity)


_TS_UNITS = {"MICROSECOND": 1, "MILLISECOND": 1000, "SECOND": 1_000_000,
             "MINUTE": 60_000_000, "HOUR": 3_600_000_000,
             "DAY": 86_400_000_000, "WEEK": 7 * 86_400_000_000}


def _ts_us(c):
    us = c.data.to(torch.int64)
    if isinstance(c.dtype, T.DateType):
        us = us * 86_400_000_000
    return us


def _f_timestampadd(args, out, chunk, ev):
    unit = str(_scalarize(args[0]).value).upper()
    n = _col(args[1], chunk).data.to(torch.int64)
    c = _col(args[2], chunk)
    us = _ts_us(c)
    if unit in _TS_UNITS:
        res = us + n * _TS_UNITS[unit]
    elif unit in ("MONTH", "QUARTER", "YEAR"):
        import datetime as _dt

        mult = {"MONTH": 1, "QUARTER": 3, "YEAR": 12}[unit]
        vals = []
        for u, k in zip(us.tolist(), n.tolist()):
            d = _dt.datetime(1970, 1, 1) + _dt.timedelta(microseconds=u)
            total = d.year * 12 + (d.month - 1) + k * mult
            y, m = divmod(total, 12)
            import calendar as _cal

            day = min(d.day, _cal.monthrange(y, m + 1)[1])
            nd = d.replace(year=y, month=m + 1, day=day)
            vals.append(int((nd - _dt.datetime(1970, 1, 1)).total_seconds()
                            * 1_000_000) + d.microsecond % 1)
        res = torch.tensor(vals, dtype=torch.int64, device=chunk.device)
    else:
        raise ValueError(f"timestampadd: bad unit {unit}")
    return Column(T.TIMESTAMP, res, c.validity)


def _f_timestampdiff(args, out, chunk, ev):
    unit = str(_scalarize(args[0]).value).upper()
    a = _col(args[1], chunk)
    b = _col(args[2], chunk)
    ua, ub = _ts_us(a), _ts_us(b)
    if unit in _TS_UNITS:
        res = torch.div(ub - ua, _TS_UNITS[unit], rounding_mode="trunc")
    elif unit in ("MONTH", "QUARTER", "YEAR"):
        import datetime as _dt

        div = {"MONTH": 1, "QUARTER": 3, "YEAR": 12}[unit]
        vals = []
        for x, y in zip(ua.tolist(), ub.tolist()):
            da = _dt.datetime(1970, 1, 1) + _dt.timedelta(microseconds=x)
            db = _dt.datetime(1970, 1, 1) + _dt.timedelta(microseconds=y)
            months = (db.year - da.year) * 12 + db.month - da.month
            # partial months don't count (Spark truncates toward zero)
            if months > 0 and (db.day, db.time()) < (da.day, da.time()):
                months -= 1
            elif months < 0 and (db.day, db.time()) > (da.day, da.time()):
                months += 1
            vals.append(months // div if months >= 0 else -((-months) // div))
        res = torch.tensor(vals, dtype=torch.int64, device=chunk.device)
    else:
        raise ValueError(f"timestampdiff: bad unit {unit}")
    validity = None
    if a.validity is not None or b.validity is not None:
        validity = (a.valid_mask() & b.valid_mask()).to(torch.uint8)
    return Column(T.I64, res, validity)


def _f_convert_timezone(args, out, chunk, ev):
    from zoneinfo import ZoneInfo

    import datetime as _dt

    if len(args) == 2:
        src, tgt, col = "UTC", _scalarize(args[0]).value, args[1]
    else:
        src, tgt, col = (_scalarize(args[0]).value,
                         _scalarize(args[1]).value, args[2])
    c = _col(col, chunk)
    zsrc, ztgt = ZoneInfo(src), ZoneInfo(tgt)
    vals = []
    for u in _ts_us(c).tolist():
        d = _dt.datetime(1970, 1, 1) + _dt.timedelta(microseconds=u)
        d2 = d.replace(tzinfo=zsrc).astimezone(ztgt).replace(tzinfo=None)
        vals.append(int((d2 - _dt.datetime(1970, 1, 1)).total_seconds()
                        * 1_000_000))
    return Column(T.TIMESTAMP,
                  torch.tensor(vals, dtype=torch.int64, device=chunk.device),
                  c.validity)


def _f_getbit(args, out, chunk, ev):
    c = _col(args[0], chunk)
    from .eval import broadcast

    pos = broadcast(args[1], chunk.num_rows, chunk.device)
    res = (c.data.to(torch.int64) >> pos.data.to(torch.int64)) & 1
    validity = None
    if c.validity is not None or pos.validity is not None:
        validity = (c.valid_mask() & pos.valid_mask()).to(torch.uint8)
    return Column(T.I32, res.to(torch.int32), validity)


def _f_random(args, out, chunk, ev):
    n = chunk.num_rows
    if args:
        g = torch.Generator(device="cpu")
        g.manual_seed(int(_scalarize(args[0]).value))
        vals = torch.rand(n, generator=g, dtype=torch.float64).to(chunk.device)
    else:
        vals = torch.rand(n, dtype=torch.float64, device=chunk.device)
    return Column(T.F64, vals, None)


def _f_uniform(args, out, chunk, ev):
    lo = int(_scalarize(args[0]).value)
    hi = int(_scalarize(args[1]).value)
    n = chunk.num_rows
    if len(args) > 2:
        g = torch.Generator(device="cpu")
        g.manual_seed(int(_scalarize(args[2]).value))
        vals = torch.randint(lo, hi + 1, (n,), generator=g).to(chunk.device)
    else:
        vals = torch.randint(lo, hi + 1, (n,), device=chunk.device)
    return Column(T.I64, vals.to(torch.int64), None)


def _f_randstr(args, out, chunk, ev):
    import random as _random
    import string as _string

    from .column import StringColumn

    ln = int(_scalarize(args[0]).value)
    rng = _random.Random(int(_scalarize(args[1]).value)) \
        if len(args) > 1 else _random
    alpha = _string.ascii_letters + _string.digits
    vals = ["".join(rng.choice(alpha) for _ in range(ln))
            for _ in range(chunk.num_rows)]
    return StringColumn.from_pylist(vals, device=str(chunk.device),
                                    dict_encode=False)


def _to_number_one(v, fmt):
    if v is None:
        return None
    t = v.strip().replace(",", "").replace("$", "")
    neg = t.startswith("-") or (t.startswith("(") and t.endswith(")"))
    t = t.strip("()-+")
    try:
        x = float(t)
    except ValueError:
        return None
    return -x if neg else x


def _f_to_number(args, out, chunk, ev):
    fmt = _scalarize(args[1]).value if len(args) > 1 else None
    c = _col(args[0], chunk)
    vals = [_to_number_one(v, fmt) for v in c.to_pylist()]
    raws = c.to_pylist()
    bad = [r for r, v in zip(raws, vals) if r is not None and v is None]
    if bad:
        raise ValueError(f"to_number: cannot parse {bad[0]!r}")
    return Column.from_values(vals, out, device=str(chunk.device))


def _f_try_to_number(args, out, chunk, ev):
    fmt = _scalarize(args[1]).value if len(args) > 1 else None
    c = _col(args[0], chunk)
    vals = [_to_number_one(v, fmt) for v in c.to_pylist()]
    return Column.from_values(vals, out, device=str(chunk.device))


def _to_binary_one(v, fmt, strict):
    import base64 as _b64

    if v is None:
        return None
    try:
        if fmt == "utf-8" or fmt == "utf8":
            return v
        if fmt == "hex":
            return bytes.fromhex(v).decode("latin-1")
        if fmt == "base64":
            return _b64.b64decode(v).decode("latin-1")
    except (ValueError, TypeError):
        if strict:
            raise ValueError(f"to_binary: cannot decode {v!r} as {fmt}")
        return None
    raise ValueError(f"to_binary: unknown format {fmt!r}")


def _f_to_binary(args, out, chunk, ev):
    fmt = (str(_scalarize(args[1]).value).lower()
           if len(args) > 1 else "hex")
    return _dict_transform(
        lambda v: _to_binary_one(v, fmt, True))(args[:1], out, chunk, ev)


def _f_try_to_binary(args, out, chunk, ev):
    fmt = (str(_scalarize(args[1]).value).lower()
           if len(args) > 1 else "hex")
    return _dict_transform(
        lambda v: _to_binary_one(v, fmt, False))(args[:1], out, chunk, ev)


def _f_to_varchar(args, out, chunk, ev):
    from .column import StringColumn

    c = _col(args[0], chunk)
    vals = c.to_pylist()
    res = [None if v is None else
           ("true" if v is True else "false" if v is False else str(v))
           for v in vals]
    return StringColumn.from_pylist(res, device=str(chunk.device))


def _f_try_mod(args, out, chunk, ev):
    from .eval import broadcast

    n = chunk.num_rows
    a = broadcast(args[0], n, chunk.device)
    b = broadcast(args[1], n, chunk.device)
    bz = b.data == 0
    safe_b = torch.where(bz, torch.ones_like(b.data), b.data)
    res = torch.fmod(a.data, safe_b) if a.data.is_floating_point() \
        else a.data - torch.div(a.data, safe_b, rounding_mode="trunc") * safe_b
    valid = a.valid_mask() & b.valid_mask() & ~bz
    return Column(out or a.dtype, res, valid.to(torch.uint8))


def _f_current_schema(args, out, chunk, ev):
    from .column import StringColumn

    return StringColumn.from_pylist(["default"] * chunk.num_rows,
                                    device=str(chunk.device))


def _f_user(args, out, chunk, ev):
    import getpass

    from .column import StringColumn

    try:
        u = getpass.getuser()
    except Exception:
        u = "unknown"
    return StringColumn.from_pylist([u] * chunk.num_rows,
                                    device=str(chunk.device))


_IMPLS["strpos"] = _f_strpos
_IMPLS["quote"] = _f_quote
_IMPLS["mask"] = _f_mask
_IMPLS["regexp_count"] = _f_regexp_count
_IMPLS["regexp_instr"] = _f_regexp_instr
_IMPLS["regexp_substr"] = _f_regexp_substr
_IMPLS["dayname"] = _f_dayname
_IMPLS["date_from_unix_date"] = _f_date_from_unix_date
_IMPLS["timestampadd"] = _f_timestampadd
_IMPLS["timestampdiff"] = _f_timestampdiff
_IMPLS["timestamp_add"] = _f_timestampadd
_IMPLS["timestamp_diff"] = _f_timestampdiff
_IMPLS["convert_timezone"] = _f_convert_timezone
_IMPLS["getbit"] = _f_getbit
_IMPLS["bit_get"] = _f_getbit
_IMPLS["random"] = _f_random
_IMPLS["uniform"] = _f_uniform
_IMPLS["randstr"] = _f_randstr
_IMPLS["to_number"] = _f_to_number
_IMPLS["try_to_number"] = _f_try_to_number
_IMPLS["to_binary"] = _f_to_binary
_IMPLS["try_to_binary"] = _f_try_to_binary
_IMPLS["to_varchar"] = _f_to_varchar
_IMPLS["to_char"] = _f_to_varchar
_IMPLS["try_mod"] = _f_try_mod
_IMPLS["current_schema"] = _f_current_schema
_IMPLS["user"] = _f_user
_IMPLS["session_user"] = _f_user


# -- XML struct parsing + collation stubs ----------------------------------

def _f_from_xml(args, out, chunk, ev):
    """from_xml(col, 'a INT, b STRING') -> struct: each field read from the
    matching child element of the row's root (ref: sail-function
    scalar/xml)."""
    import xml.etree.ElementTree as ET

    from .column import StructColumn

    c = _col(args[0], chunk)
    docs = c.to_pylist()
    parsed = []
    for d in docs:
        if d is None:
            parsed.append(None)
            continue
        try:
            parsed.append(ET.fromstring(d))
        except ET.ParseError:
            parsed.append(None)
    kids = []
    for f in out.fields:
        vals = []
        for root in parsed:
            if root is None:
                vals.append(None)
                continue
            el = root.find(f.name)
            txt = None if el is None else (el.text or "")
            if txt is None or isinstance(f.dtype, T.StringType):
                vals.append(txt)
            else:
                try:
                    vals.append(float(txt) if isinstance(
                        f.dtype, (T.Float32Type, T.Float64Type,
                                  T.DecimalType)) else int(txt))
                except ValueError:
                    vals.append(None)
        if isinstance(f.dtype, T.StringType):
            kids.append((f.name, StringColumn.from_pylist(
                vals, device=str(chunk.device))))
        else:
            kids.append((f.name, Column.from_values(
                vals, f.dtype, device=chunk.device)))
    validity = None
    if any(p is None for p in parsed):
        validity = torch.tensor([0 if p is None else 1 for p in parsed],
                                dtype=torch.uint8, device=chunk.device)
    return StructColumn(kids, validity, dtype=out)


def _f_to_xml(args, out, chunk, ev):
    from .column import StringColumn
    from xml.sax.saxutils import escape

    c = _col(args[0], chunk)
    names = [nm for nm, _ in c.children_]
    cols = [kid.to_pylist() for _, kid in c.children_]
    parts = []
    for i in range(len(c)):
        fields = []
        for nm, vals in zip(names, cols):
            v = vals[i]
            fields.append(f"<{nm}/>" if v is None
                          else f"<{nm}>{escape(str(v))}</{nm}>")
        parts.append("<ROW>" + "".join(fields) + "</ROW>")
    col = StringColumn.from_pylist(parts, device=str(chunk.device))
    col.validity = c.validity
    return col


def _f_schema_of_xml(args, out, chunk, ev):
    import xml.etree.ElementTree as ET

    from .column import StringColumn

    sample = _scalarize(args[0]).value
    root = ET.fromstring(sample)
    fields = []
    for el in root:
        txt = (el.text or "").strip()
        try:
            int(txt)
            t = "BIGINT"
        except ValueError:
            try:
                float(txt)
                t = "DOUBLE"
            except ValueError:
                t = "STRING"
        fields.append(f"{el.tag}: {t}")
    body = ", ".join(fields)
    return StringColumn.from_pylist([f"STRUCT<{body}>"] * chunk.num_rows,
                                    device=str(chunk.device))


def _f_collate(args, out, chunk, ev):
    # single-collation engine (UTF8_BINARY): collate() is identity
    return _col(args[0], chunk)


def _f_collation(args, out, chunk, ev):
    from .column import StringColumn

    return StringColumn.from_pylist(["UTF8_BINARY"] * chunk.num_rows,
                                    device=str(chunk.device))


_IMPLS["from_xml"] = _f_from_xml
_IMPLS["to_xml"] = _f_to_xml
_IMPLS["schema_of_xml"] = _f_schema_of_xml
_IMPLS["collate"] = _f_collate
_IMPLS["collation"] = _f_collation


def _f_make_dt_interval(args, out, chunk, ev):
    """make_dt_interval([days, hours, mins, secs]) -> day-time interval,
    represented as int64 microseconds (the engine's interval storage —
    adds directly to TIMESTAMP columns)."""
    from .eval import broadcast

    n = chunk.num_rows
    mults = [86_400_000_000, 3_600_000_000, 60_000_000, 1_000_000]
    total = torch.zeros(n, dtype=torch.int64, device=chunk.device)
    validity = None
    for i, m in enumerate(mults):
        if len(args) <= i:
            break
        c = broadcast(args[i], n, chunk.device)
        if i == 3 and c.data.is_floating_point():
            total = total + (c.data.to(torch.float64) * m).to(torch.int64)
        else:
            total = total + c.data.to(torch.int64) * m
        if c.validity is not None:
            v = c.valid_mask()
            validity = v if validity is None else (validity & v)
    return Column(T.I64, total,
                  validity.to(torch.uint8) if validity is not None else None)


_IMPLS["make_dt_interval"] = _f_make_dt_interval


# -- last trivial parity batch ---------------------------------------------

def _f_trig_recip(fn):
    def run(args, out, chunk, ev):
        c = _col(args[0], chunk)
        return Column(T.F64, 1.0 / fn(c.data.to(torch.float64)), c.validity)
    return run


def _f_current_timezone(args, out, chunk, ev):
    from .column import StringColumn

    return StringColumn.from_pylist(["UTC"] * chunk.num_rows,
                                    device=str(chunk.device))


def _f_true_const(args, out, chunk, ev):
    # engine strings are python str: always valid UTF-8 by construction
    return Column(T.BOOL, torch.ones(chunk.num_rows, dtype=torch.bool,
                                     device=chunk.device), None)


def _f_identity_str(args, out, chunk, ev):
    return _col(args[0], chunk)


def _f_is_valid_variant(args, out, chunk, ev):
    import json as _json

    c = _col(args[0], chunk)

    def ok(v):
        if v is None:
            return None
        try:
            _json.loads(v)
            return True
        except (ValueError, TypeError):
            return False

    return Column.from_values([ok(v) for v in c.to_pylist()], T.BOOL,
                              device=str(chunk.device))


def _f_random_poisson(args, out, chunk, ev):
    lam = float(_scalarize(args[0]).value) if args else 1.0
    vals = torch.poisson(torch.full((chunk.num_rows,), lam,
                                    dtype=torch.float64, device=chunk.device))
    return Column(T.I64, vals.to(torch.int64), None)


def _f_bitmap_bit_position(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return Column(T.I64, (c.data.to(torch.int64) - 1) % 32768, c.validity)


def _f_bitmap_bucket_number(args, out, chunk, ev):
    c = _col(args[0], chunk)
    return Column(T.I64,
                  torch.div(c.data.to(torch.int64) - 1, 32768,
                            rounding_mode="floor") + 1, c.validity)


def _f_deep_size(args, out, chunk, ev):
    import sys

    c = _col(args[0], chunk)
    vals = [None if v is None else sys.getsizeof(v) for v in c.to_pylist()]
    return Column.from_values(vals, T.I64, device=str(chunk.device))


_IMPLS["cot"] = _f_trig_recip(torch.tan)
_IMPLS["csc"] = _f_trig_recip(torch.sin)
_IMPLS["sec"] = _f_trig_recip(torch.cos)
_IMPLS["current_timezone"] = _f_current_timezone
_IMPLS["is_valid_utf8"] = _f_true_const
_IMPLS["try_validate_utf8"] = _f_identity_str
_IMPLS["validate_utf8"] = _f_identity_str
_IMPLS["make_valid_utf8"] = _f_identity_str
_IMPLS["variant_to_json"] = _f_identity_str
_IMPLS["to_variant_object"] = _f_parse_json
_IMPLS["is_valid_variant"] = _f_is_valid_variant
_IMPLS["try_url_decode"] = _f_url_decode
_IMPLS["random_poisson"] = _f_random_poisson
_IMPLS["binary"] = _f_to_varchar
_IMPLS["bitmap_bit_position"] = _f_bitmap_bit_position
_IMPLS["bitmap_bucket_number"] = _f_bitmap_bucket_number
_IMPLS["deep_size"] = _f_deep_size


def _f_ts_add_months(args, out, chunk, ev):
    from .eval import Scalar, broadcast

    n = chunk.num_rows
    months = broadcast(args[1], n, chunk.device).data.to(torch.int64)
    unit = Scalar("MONTH", T.STRING)
    return _f_timestampadd([unit, Column(T.I64, months, None), args[0]],
                           out, chunk, ev)


_IMPLS["ts_add_months"] = _f_ts_add_months

from . import functions_ext  # noqa: E402,F401  (registers the round-2 batch)
