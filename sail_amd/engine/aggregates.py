"""Group-by aggregation (torch path).

Reference implementation of the ~sail-function aggregate surface
(ref: crates/sail-function/src/aggregate/) over dense group ids; the GPU hot
path is the HIP hash-aggregate kernel (ops/csrc/hash_agg.hip) validated
against this module.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from . import types as T
from .column import Column, StringColumn
from .joins import normalize_key


def group_ids(keys: List[Column]) -> Tuple[torch.Tensor, torch.Tensor, int]:
    """Returns (gid per row, representative row index per group, n_groups).
    Null keys form their own groups (SQL GROUP BY semantics)."""
    if not keys:
        n = 0
        raise ValueError("group_ids requires keys")
    dev = keys[0].device
    n = len(keys[0])
    norm = []
    for c in keys:
        k = normalize_key(c)
        if c.validity is not None:
            # give nulls a dedicated code below the domain
            k = torch.where(c.valid_mask(), k, torch.full_like(k, k.min().item() - 1 if n else -1))
        norm.append(k)
    if len(norm) == 1:
        uniq, gid = torch.unique(norm[0], return_inverse=True)
        ng = int(uniq.shape[0])
    else:
        stacked = torch.stack(norm, dim=1)
        uniq, gid = torch.unique(stacked, dim=0, return_inverse=True)
        ng = int(uniq.shape[0])
    # representative row per group (first occurrence for determinism)
    rep = torch.full((ng,), n, dtype=torch.int64, device=dev)
    rep.scatter_reduce_(0, gid, torch.arange(n, device=dev), reduce="amin", include_self=True)
    return gid, rep, ng


def _masked(values: Column, extra_mask: Optional[torch.Tensor]):
    mask = values.valid_mask()
    if extra_mask is not None:
        mask = mask & extra_mask
    return mask


def agg_eval(name: str, args: List[Column], gid: torch.Tensor, ng: int,
             distinct: bool = False, filter_mask: Optional[torch.Tensor] = None,
             out_type: T.DataType = None) -> Column:
    """Evaluate one aggregate over groups. args already evaluated per-row."""
    dev = gid.device
    if name == "count" and not args:
        mask = filter_mask if filter_mask is not None else torch.ones(gid.shape[0], dtype=torch.bool, device=dev)
        data = torch.zeros(ng, dtype=torch.int64, device=dev)
        data.index_add_(0, gid[mask], torch.ones(int(mask.sum()), dtype=torch.int64, device=dev))
        return Column(T.I64, data, None)

    c = args[0]
    mask = _masked(c, filter_mask)
    if distinct and name in ("count", "sum", "avg"):
        # reduce (gid, value) pairs to unique before aggregating
        vk = normalize_key(c)
        pair = torch.stack([gid[mask], vk[mask]], dim=1)
        upair = torch.unique(pair, dim=0)
        ugid = upair[:, 0]
        if name == "count":
            data = torch.zeros(ng, dtype=torch.int64, device=dev)
            data.index_add_(0, ugid, torch.ones(ugid.shape[0], dtype=torch.int64, device=dev))
            return Column(T.I64, data, None)
        # sum/avg distinct: gather original values — only valid for int-like
        vals = upair[:, 1]
        data = torch.zeros(ng, dtype=torch.int64, device=dev)
        data.index_add_(0, ugid, vals)
        if name == "avg":
            cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
            cnt.index_add_(0, ugid, torch.ones(ugid.shape[0], dtype=torch.int64, device=dev))
            return _avg_result(data, cnt, c.dtype, out_type)
        return Column(out_type or T.I64, data, None)

    gidm = gid[mask]
    n_used = int(mask.sum().item())

    if name == "count":
        data = torch.zeros(ng, dtype=torch.int64, device=dev)
        data.index_add_(0, gidm, torch.ones(n_used, dtype=torch.int64, device=dev))
        return Column(T.I64, data, None)

    if name == "sumf":
        vals = c.data[mask].to(torch.float64)
        if isinstance(c.dtype, T.DecimalType):
            vals = vals / (10.0 ** c.dtype.scale)
        data = torch.zeros(ng, dtype=torch.float64, device=dev)
        data.index_add_(0, gidm, vals)
        return Column(T.F64, data, None)

    if name == "sumsq":
        vals = c.data[mask].to(torch.float64)
        if isinstance(c.dtype, T.DecimalType):
            vals = vals / (10.0 ** c.dtype.scale)
        data = torch.zeros(ng, dtype=torch.float64, device=dev)
        data.index_add_(0, gidm, vals * vals)
        return Column(T.F64, data, None)

    if name == "count_if":
        vals = c.data[mask].to(torch.int64)
        data = torch.zeros(ng, dtype=torch.int64, device=dev)
        data.index_add_(0, gidm, vals)
        return Column(T.I64, data, None)

    if name in ("sum", "try_sum"):
        if isinstance(c.dtype, T.DecimalType) or c.dtype.is_integer:
            vals = c.data[mask].to(torch.int64)
            acc_t = torch.int64
        else:
            vals = c.data[mask].to(torch.float64)
            acc_t = torch.float64
        data = torch.zeros(ng, dtype=acc_t, device=dev)
        data.index_add_(0, gidm, vals)
        cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
        cnt.index_add_(0, gidm, torch.ones(n_used, dtype=torch.int64, device=dev))
        validity = (cnt > 0).to(torch.uint8)
        rt = out_type or (T.DecimalType(38, c.dtype.scale) if isinstance(c.dtype, T.DecimalType)
                          else (T.I64 if c.dtype.is_integer else T.F64))
        if isinstance(rt, T.DecimalType) and isinstance(c.dtype, T.DecimalType) and rt.scale != c.dtype.scale:
            from .eval import _rescale_int

            data = _rescale_int(data, c.dtype.scale, rt.scale)
        return Column(rt, data, None if bool((cnt > 0).all()) else validity)

    if name in ("avg", "try_avg"):
        if isinstance(c.dtype, T.DecimalType) or c.dtype.is_integer:
            vals = c.data[mask].to(torch.int64)
            data = torch.zeros(ng, dtype=torch.int64, device=dev)
        else:
            vals = c.data[mask].to(torch.float64)
            data = torch.zeros(ng, dtype=torch.float64, device=dev)
        data.index_add_(0, gidm, vals)
        cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
        cnt.index_add_(0, gidm, torch.ones(n_used, dtype=torch.int64, device=dev))
        return _avg_result(data, cnt, c.dtype, out_type)

    if name in ("min", "max"):
        red = "amin" if name == "min" else "amax"
        if isinstance(c, StringColumn):
            # order by dict code only if dictionary is sorted; our dictionaries
            # are built sorted (column.py from_pylist), so codes order == value order
            vals = normalize_key(c)[mask]
            data = torch.full((ng,), 2**62 if name == "min" else -(2**62), dtype=torch.int64, device=dev)
            data.scatter_reduce_(0, gidm, vals, reduce=red, include_self=True)
            cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
            cnt.index_add_(0, gidm, torch.ones(n_used, dtype=torch.int64, device=dev))
            if c.is_dict:
                return StringColumn(c.offsets, c.bytes_,
                                    None if bool((cnt > 0).all()) else (cnt > 0).to(torch.uint8),
                                    data.to(torch.int32))
            raise NotImplementedError("min/max over raw strings TODO")
        vals = c.data[mask]
        if vals.dtype.is_floating_point:
            init = float("inf") if name == "min" else float("-inf")
        else:
            info = torch.iinfo(vals.dtype)
            init = info.max if name == "min" else info.min
        data = torch.full((ng,), init, dtype=vals.dtype, device=dev)
        data.scatter_reduce_(0, gidm, vals, reduce=red, include_self=True)
        cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
        cnt.index_add_(0, gidm, torch.ones(n_used, dtype=torch.int64, device=dev))
        return Column(out_type or c.dtype, data,
                      None if bool((cnt > 0).all()) else (cnt > 0).to(torch.uint8))

    if name in ("first", "last", "any_value"):
        sel = torch.arange(gid.shape[0], device=dev)[mask]
        init = gid.shape[0] if name != "last" else -1
        red = "amin" if name != "last" else "amax"
        rep = torch.full((ng,), init, dtype=torch.int64, device=dev)
        rep.scatter_reduce_(0, gidm, sel, reduce=red, include_self=True)
        ok = rep != init
        safe = torch.where(ok, rep, torch.zeros_like(rep))
        res = c.gather(safe)
        if not bool(ok.all()):
            v = res.valid_mask() & ok
            if isinstance(res, StringColumn):
                res.validity = v.to(torch.uint8)
            else:
                res = Column(res.dtype, res.data, v.to(torch.uint8))
        return res

    if name in ("stddev_samp", "stddev_pop", "var_samp", "var_pop"):
        vals = c.data[mask].to(torch.float64)
        if isinstance(c.dtype, T.DecimalType):
            vals = vals / (10.0 ** c.dtype.scale)
        s1 = torch.zeros(ng, dtype=torch.float64, device=dev)
        s2 = torch.zeros(ng, dtype=torch.float64, device=dev)
        cnt = torch.zeros(ng, dtype=torch.float64, device=dev)
        s1.index_add_(0, gidm, vals)
        s2.index_add_(0, gidm, vals * vals)
        cnt.index_add_(0, gidm, torch.ones(n_used, dtype=torch.float64, device=dev))
        mean = s1 / cnt.clamp_min(1)
        m2 = s2 - cnt * mean * mean
        denom = cnt - (1.0 if name.endswith("_samp") else 0.0)
        var = m2 / denom.clamp_min(1e-300)
        var = var.clamp_min(0)
        data = torch.sqrt(var) if name.startswith("stddev") else var
        valid = denom > 0
        return Column(T.F64, data, None if bool(valid.all()) else valid.to(torch.uint8))

    if name in ("any", "bool_and"):
        vals = c.data[mask].to(torch.bool)
        if name == "any":
            data = torch.zeros(ng, dtype=torch.bool, device=dev)
            data.scatter_reduce_(0, gidm, vals, reduce="amax", include_self=True)
        else:
            data = torch.ones(ng, dtype=torch.bool, device=dev)
            data.scatter_reduce_(0, gidm, vals, reduce="amin", include_self=True)
        return Column(T.BOOL, data, None)

    if name == "product":
        vals = c.data[mask].to(torch.float64)
        data = torch.ones(ng, dtype=torch.float64, device=dev)
        data.scatter_reduce_(0, gidm, vals, reduce="prod", include_self=True)
        return Column(T.F64, data, None)

    if name in ("median", "percentile", "percentile_approx"):
        # exact median/percentile via host (small group counts expected)
        vals = c.data[mask].to(torch.float64)
        if isinstance(c.dtype, T.DecimalType):
            vals = vals / (10.0 ** c.dtype.scale)
        out = torch.zeros(ng, dtype=torch.float64, device=dev)
        g = gidm
        order = torch.argsort(g * (2 ** 20) + torch.argsort(torch.argsort(vals)))
        # simple per-group via host loop fallback (reference correctness path)
        import numpy as np

        gnp = g.cpu().numpy()
        vnp = vals.cpu().numpy()
        q = 0.5
        import collections

        groups = collections.defaultdict(list)
        for gg, vv in zip(gnp, vnp):
            groups[int(gg)].append(vv)
        res = np.zeros(ng)
        for k, lst in groups.items():
            res[k] = float(np.percentile(lst, q * 100))
        return Column(T.F64, torch.from_numpy(res).to(dev), None)

    raise NotImplementedError(f"aggregate {name}")


def _avg_result(sums: torch.Tensor, cnt: torch.Tensor, in_type: T.DataType, out_type) -> Column:
    dev = sums.device
    valid = cnt > 0
    safe = cnt.clamp_min(1)
    if isinstance(in_type, T.DecimalType):
        ot = out_type if isinstance(out_type, T.DecimalType) else T.DecimalType(38, min(in_type.scale + 4, 10))
        # sums at in.scale; result at ot.scale
        num = sums.to(torch.float64) * (10.0 ** (ot.scale - in_type.scale))
        data = torch.round(num / safe.to(torch.float64)).to(torch.int64)
        return Column(ot, data, None if bool(valid.all()) else valid.to(torch.uint8))
    data = sums.to(torch.float64) / safe.to(torch.float64)
    return Column(T.F64, data, None if bool(valid.all()) else valid.to(torch.uint8))


def global_ids(n: int, device) -> Tuple[torch.Tensor, int]:
    """gid tensor for a global (no GROUP BY) aggregate: all rows in group 0."""
    return torch.zeros(n, dtype=torch.int64, device=device), 1
