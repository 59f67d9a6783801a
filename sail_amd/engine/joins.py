"""Join algorithms (torch path).

Sort/search-based equi-join used as the engine's reference implementation and
CPU path; the GPU path swaps in the HIP open-addressing hash join
(ops/csrc/hash_join.hip) with identical semantics. All Spark join types
(ref: crates/sail-common/src/spec/plan.rs:1148 JoinType).

Keys are normalized to int64 "dense ids" via torch.unique over the
concatenated build+probe key columns, so multi-column and string keys join
exactly (no hash collisions on this path).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from . import types as T
from .column import Column, StringColumn


def normalize_key(c: Column) -> torch.Tensor:
    """Map a key column to int64 where equal values have equal codes.
    Nulls map to a sentinel that never matches (handled by caller masks)."""
    if isinstance(c, StringColumn):
        if c.is_dict:
            return c.codes.to(torch.int64)
        # raw strings: host-side dictionary encode (CPU reference path)
        vals = c.to_pylist()
        lut = {}
        out = []
        for v in vals:
            if v is None:
                out.append(-1)
            else:
                out.append(lut.setdefault(v, len(lut)))
        return torch.tensor(out, dtype=torch.int64, device=c.device)
    if c.data.dtype == torch.float64:
        return c.data.view(torch.int64)
    if c.data.dtype == torch.float32:
        return c.data.view(torch.int32).to(torch.int64)
    return c.data.to(torch.int64)


def dense_ids(build_keys: List[torch.Tensor], probe_keys: List[torch.Tensor]):
    """Assign identical int64 ids to identical key tuples across both sides."""
    nb = build_keys[0].shape[0]
    if len(build_keys) == 1:
        allk = torch.cat([build_keys[0], probe_keys[0]])
        _, inv = torch.unique(allk, return_inverse=True)
    else:
        allk = torch.stack([torch.cat([b, p]) for b, p in zip(build_keys, probe_keys)], dim=1)
        _, inv = torch.unique(allk, dim=0, return_inverse=True)
    return inv[:nb], inv[nb:]


def _expand_matches(bids: torch.Tensor, pids: torch.Tensor,
                    b_valid: Optional[torch.Tensor], p_valid: Optional[torch.Tensor]):
    """Core equi-match: returns (probe_idx, build_idx) pairs for all matches,
    plus per-probe match counts and build-side matched flags."""
    nb, np_ = bids.shape[0], pids.shape[0]
    dev = bids.device
    if b_valid is not None:
        bids = torch.where(b_valid, bids, torch.full_like(bids, -1))
    if p_valid is not None:
        pids = torch.where(p_valid, pids, torch.full_like(pids, -2))
    order = torch.argsort(bids)
    bsorted = bids[order]
    lo = torch.searchsorted(bsorted, pids, right=False)
    hi = torch.searchsorted(bsorted, pids, right=True)
    counts = (hi - lo).clamp_min(0)
    total = int(counts.sum().item())
    probe_idx = torch.repeat_interleave(torch.arange(np_, device=dev), counts)
    if total:
        cum = torch.zeros(np_, dtype=torch.int64, device=dev)
        torch.cumsum(counts, 0, out=cum)
        excl = cum - counts
        pos = torch.arange(total, device=dev)
        within = pos - excl[probe_idx]
        build_idx = order[lo[probe_idx] + within]
    else:
        build_idx = torch.zeros(0, dtype=torch.int64, device=dev)
    return probe_idx, build_idx, counts


def equi_join(build_keys: List[Column], probe_keys: List[Column], how: str,
              ) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """Returns (probe_indices, build_indices, probe_unmatched_mask).

    how semantics are relative to (build=right side? no): caller orients.
    - "inner": matched pairs only
    - "left_rows": pairs + unmatched probe rows marked (for outer fill)
    - "semi"/"anti": probe filtering handled by caller with counts
    """
    bk = [normalize_key(c) for c in build_keys]
    pk = [normalize_key(c) for c in probe_keys]
    bids, pids = dense_ids(bk, pk)
    bv = None
    pv = None
    for c in build_keys:
        if c.validity is not None:
            m = c.valid_mask()
            bv = m if bv is None else (bv & m)
    for c in probe_keys:
        if c.validity is not None:
            m = c.valid_mask()
            pv = m if pv is None else (pv & m)
    probe_idx, build_idx, counts = _expand_matches(bids, pids, bv, pv)
    return probe_idx, build_idx, counts
