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
        return raw_string_key(c)
    if c.data.dtype == torch.float64:
        return c.data.view(torch.int64)
    if c.data.dtype == torch.float32:
        return c.data.view(torch.int32).to(torch.int64)
    return c.data.to(torch.int64)


def normalize_key_pair(a: Column, b: Column):
    """Normalize a pair of join-key columns to comparable int64 keys.
    Raw-string pairs must share one key scheme (packed vs hashed)."""
    if isinstance(a, StringColumn) and isinstance(b, StringColumn) \
            and not a.is_dict and not b.is_dict:
        la = a.offsets[1:] - a.offsets[:-1]
        lb = b.offsets[1:] - b.offsets[:-1]
        ma = int(la.max().item()) if len(a) else 0
        mb = int(lb.max().item()) if len(b) else 0
        if max(ma, mb) <= 7:
            return raw_string_key(a), raw_string_key(b)
        # exact shared codes across both sides (no FNV-collision class):
        # equal strings get equal codes, distinct strings never do
        ka, kb = exact_string_codes([a, b])
        return ka, kb
    if isinstance(a, StringColumn) and isinstance(b, StringColumn) \
            and a.is_dict != b.is_dict:
        # one side dict, one raw: exact codes over (dict values, raw rows);
        # the dict side maps its per-row codes through the value codes
        da = a if a.is_dict else b
        ra = b if a.is_dict else a
        la = ra.offsets[1:] - ra.offsets[:-1]
        force = (int(la.max().item()) if len(ra) else 0) > 7 or \
            max((len(v) for v in da.dict_values()), default=0) > 7
        if not force:
            dk = _dict_side_key(da, False)
            rk = raw_string_key(ra, False)
            return (dk, rk) if a.is_dict else (rk, dk)
        dvals = StringColumn(da.offsets, da.bytes_, None, None)
        vcodes, rk = exact_string_codes([dvals, ra])
        dk = vcodes.index_select(0, da.codes.to(torch.int64).clamp_min(0))
        return (dk, rk) if a.is_dict else (rk, dk)
    return normalize_key(a), normalize_key(b)


def _dict_side_key(c: StringColumn, force_hash: bool) -> torch.Tensor:
    from .eval_keys import literal_keys_like

    vals = c.dict_values()
    lut = literal_keys_like(vals, force_hash, c.device)
    key = lut[c.codes.to(torch.int64).clamp_min(0)]
    return torch.where(c.codes >= 0, key, torch.full_like(key, -(10 ** 18)))


def dense_ids(build_keys: List[torch.Tensor], probe_keys: List[torch.Tensor]):
    """Comparable int64 ids for identical key tuples across both sides.

    Single-column int64 keys are used directly (no dense remap needed — the
    sort/searchsorted matcher handles arbitrary values). Multi-column keys
    try exact range-packing (k = ((k0-min0)*span1 + (k1-min1))*span2 + ...)
    and only fall back to torch.unique row-dedup when the packed domain
    overflows int64."""
    nb = build_keys[0].shape[0]
    if len(build_keys) == 1:
        return build_keys[0], probe_keys[0]
    # exact range packing
    spans = []
    mins = []
    ok = True
    for b, p in zip(build_keys, probe_keys):
        if b.numel() == 0 and p.numel() == 0:
            mins.append(0)
            spans.append(1)
            continue
        candidates = [t for t in (b, p) if t.numel()]
        lo = min(int(t.min().item()) for t in candidates)
        hi = max(int(t.max().item()) for t in candidates)
        mins.append(lo)
        spans.append(hi - lo + 1)
    total = 1
    for s in spans:
        total *= s
        if total > (1 << 62):
            ok = False
            break
    if ok:
        def pack(cols):
            acc = (cols[0] - mins[0])
            for i in range(1, len(cols)):
                acc = acc * spans[i] + (cols[i] - mins[i])
            return acc

        return pack(build_keys), pack(probe_keys)
    allk = torch.stack([torch.cat([b, p]) for b, p in zip(build_keys, probe_keys)], dim=1)
    _, inv = torch.unique(allk, dim=0, return_inverse=True)
    return inv[:nb], inv[nb:]


def _expand_matches(bids: torch.Tensor, pids: torch.Tensor,
                    b_valid: Optional[torch.Tensor], p_valid: Optional[torch.Tensor]):
    """Core equi-match: returns (probe_idx, build_idx) pairs for all matches,
    plus per-probe match counts and build-side matched flags."""
    nb, np_ = bids.shape[0], pids.shape[0]
    dev = bids.device
    # null sentinels at the extreme of the int64 domain so they can never
    # collide with real keys (keys may be arbitrary int64 now) nor each other
    if b_valid is not None:
        bids = torch.where(b_valid, bids, torch.full_like(bids, -(2 ** 63) + 1))
    if p_valid is not None:
        pids = torch.where(p_valid, pids, torch.full_like(pids, -(2 ** 63) + 2))
    if nb > 4 * max(np_, 1) and np_ > 0:
        # lopsided: hash/sort the SMALLER side regardless of join orientation
        # (q21 semi/anti probes a filtered slice against all of lineitem —
        # building the 600M-row chain table cost 26 ms/join; the pair set is
        # symmetric, so match with roles swapped and restore the contract)
        # swapped call: 1st return indexes ITS probe (= our build side),
        # 2nd indexes ITS build (= our probe side)
        build_idx, probe_idx, _ = _expand_matches(pids, bids, None, None)
        counts = torch.bincount(probe_idx, minlength=np_)
        return probe_idx, build_idx, counts
    if bids.is_cuda and nb > 0:
        from ..ops import kernels as K

        if K.available():
            return _expand_matches_gpu(bids, pids, K.require())
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
    bk, pk = [], []
    for b, p in zip(build_keys, probe_keys):
        kb, kp = normalize_key_pair(b, p)
        bk.append(kb)
        pk.append(kp)
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


# ---------------------------------------------------------------------------
# raw (non-dictionary) string keys
# ---------------------------------------------------------------------------

_FNV_OFFSET = 14695981039346656037
_FNV_PRIME = 1099511628211
_MIX = 0x9E3779B97F4A7C15
_M64 = (1 << 64) - 1


def fnv1a_hash_py(s: bytes) -> int:
    """Host mirror of ops/csrc/strings.hip string_hash64 (for literals)."""
    h = _FNV_OFFSET
    for b in s:
        h = ((h ^ b) * _FNV_PRIME) & _M64
    h ^= (len(s) * _MIX) & _M64
    return h >> 1


def short_string_key(c: StringColumn) -> torch.Tensor:
    """Pack <=7-byte strings into an exact int64 key (fixed 7-wide packing so
    keys are comparable across columns and literals)."""
    starts = c.offsets[:-1]
    lens = c.offsets[1:] - starts
    key = lens.to(torch.int64).clone()
    nb = int(c.bytes_.shape[0])
    safe_bytes = c.bytes_.to(torch.int64) if nb else None
    for k in range(7):
        if safe_bytes is None:
            key = key * 257
            continue
        idx = (starts + k).clamp(0, nb - 1)
        b = torch.where(k < lens, safe_bytes[idx], torch.zeros_like(lens))
        key = key * 257 + b
    return key


def fnv_key_tensor(c: StringColumn) -> torch.Tensor:
    """FNV-1a 64 per row, identical to the HIP kernel; CPU path is a numpy
    byte-position loop (bounded by the longest string)."""
    if c.is_cuda:
        from ..ops import kernels as K

        return K.require().string_hash64(c.offsets, c.bytes_)
    import numpy as np

    offs = c.offsets.numpy()
    byts = c.bytes_.numpy().astype(np.uint64)
    n = len(offs) - 1
    lens = offs[1:] - offs[:-1]
    max_len = int(lens.max()) if n else 0
    h = np.full(n, np.uint64(_FNV_OFFSET), dtype=np.uint64)
    starts = offs[:-1]
    prime = np.uint64(_FNV_PRIME)
    with np.errstate(over="ignore"):
        for k in range(max_len):
            live = k < lens
            idx = np.where(live, starts + k, 0)
            b = byts[idx] if byts.size else np.zeros(n, dtype=np.uint64)
            nh = (h ^ b) * prime
            h = np.where(live, nh, h)
        h = h ^ (lens.astype(np.uint64) * np.uint64(_MIX))
    return torch.from_numpy((h >> np.uint64(1)).astype(np.int64)).to(c.device)


def raw_string_key(c: StringColumn, force_hash: bool = False) -> torch.Tensor:
    """Exact packed key for <=7-byte strings; FNV-1a 64-bit beyond (same
    values on CPU and GPU). Used where keys must be VALUE-determined
    (literal comparisons, array membership). Group-by and join keys go
    through exact_string_codes instead — no collision class there."""
    lens = c.offsets[1:] - c.offsets[:-1]
    max_len = int(lens.max().item()) if len(c) else 0
    if not force_hash and max_len <= 7:
        return short_string_key(c)
    return fnv_key_tensor(c)


# -- exact string codes ------------------------------------------------------
# Two independent 64-bit hash families + byte-equality verification of
# hash-equal neighbors => exact dictionary codes for raw-string group/join
# keys (kills the FNV-collision silent-wrong-answer class; the detection of
# a (2^-128-probability) double collision falls back to host encoding).
# ref: sail-execution join planner unmatched-row discipline
# (crates/sail-execution/src/job_graph/planner.rs:101) relies on exact keys.

_H2_SEED = 0x9E3779B97F4A7C15
_H2_MULT = 0xC6A4A7935BD1E995  # murmur64A multiplier — a different family
_H2_LEN_MIX = 0x2545F4914F6CDD1D


def _signed64(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


def hash2_tensor(c: StringColumn) -> torch.Tensor:
    """Second hash family (must match string_hash64_seeded_kernel)."""
    if c.is_cuda:
        from ..ops import kernels as K

        return K.require().string_hash64_seeded(
            c.offsets, c.bytes_, _signed64(_H2_SEED), _signed64(_H2_MULT))
    import numpy as np

    offs = c.offsets.numpy()
    byts = c.bytes_.numpy().astype(np.uint64)
    n = len(offs) - 1
    lens = offs[1:] - offs[:-1]
    max_len = int(lens.max()) if n else 0
    h = np.full(n, np.uint64(_H2_SEED), dtype=np.uint64)
    starts = offs[:-1]
    mult = np.uint64(_H2_MULT)
    with np.errstate(over="ignore"):
        for k in range(max_len):
            live = k < lens
            idx = np.where(live, starts + k, 0)
            b = byts[idx] if byts.size else np.zeros(n, dtype=np.uint64)
            nh = (h ^ b) * mult
            h = np.where(live, nh, h)
        h = h ^ (lens.astype(np.uint64) * np.uint64(_H2_LEN_MIX))
    return torch.from_numpy(h.view(np.int64).copy()).to(c.device)


def string_hash_pair(c: StringColumn):
    """(h1, h2): the 128-bit combined key for exact codes. Separated out so
    tests can force collisions."""
    return fnv_key_tensor(c), hash2_tensor(c)


def str_pairs_equal(a: StringColumn, ia: torch.Tensor,
                    b: StringColumn, ib: torch.Tensor) -> torch.Tensor:
    """Byte-exact equality of (a[ia[i]], b[ib[i]]) pairs -> bool tensor."""
    if a.is_cuda:
        from ..ops import kernels as K

        return K.require().str_pairs_equal(
            a.offsets, a.bytes_, ia, b.offsets, b.bytes_, ib).to(torch.bool)
    import numpy as np

    oa = a.offsets.numpy()
    ob = b.offsets.numpy()
    ba = a.bytes_.numpy()
    bb = b.bytes_.numpy()
    out = np.zeros(ia.numel(), dtype=bool)
    for i, (x, y) in enumerate(zip(ia.numpy(), ib.numpy())):
        sa = ba[oa[x]:oa[x + 1]]
        sb = bb[ob[y]:ob[y + 1]]
        out[i] = len(sa) == len(sb) and bool((sa == sb).all())
    return torch.from_numpy(out)


def _concat_raw(cols):
    """Concatenate raw string columns into one (offsets, bytes) view."""
    if len(cols) == 1:
        return cols[0]
    dev = cols[0].device
    lens = torch.cat([c.offsets[1:] - c.offsets[:-1] for c in cols])
    offs = torch.zeros(lens.numel() + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=offs[1:])
    bytes_ = torch.cat([c.bytes_ for c in cols])
    return StringColumn(offs, bytes_, None, None)


def _host_exact_codes(comb: StringColumn) -> torch.Tensor:
    """128-bit-collision fallback: exact host dictionary encode."""
    vals = comb.to_pylist()
    seen = {}
    out = torch.empty(len(vals), dtype=torch.int64)
    for i, v in enumerate(vals):
        v = v or ""
        j = seen.get(v)
        if j is None:
            j = len(seen)
            seen[v] = j
        out[i] = j
    return out.to(comb.device)


def exact_string_codes(cols) -> list:
    """Exact dense codes for raw-string key columns: equal strings (across
    all given columns) share a code, distinct strings never do. Sort by
    (h1, h2), verify hash-equal neighbors byte-for-byte on device, assign
    codes at verified boundaries. Returns one int64 tensor per column."""
    comb = _concat_raw(cols)
    n = len(comb)
    if n == 0:
        return [torch.zeros(0, dtype=torch.int64, device=comb.device)
                for _ in cols]
    h1, h2 = string_hash_pair(comb)
    order = torch.argsort(h2, stable=True)
    order = order[torch.argsort(h1.index_select(0, order), stable=True)]
    s1 = h1.index_select(0, order)
    s2 = h2.index_select(0, order)
    same = (s1[1:] == s1[:-1]) & (s2[1:] == s2[:-1])
    codes_all = None
    idx = torch.nonzero(same, as_tuple=False).flatten()
    if idx.numel():
        ia = order.index_select(0, idx)
        ib = order.index_select(0, idx + 1)
        eq = str_pairs_equal(comb, ia, comb, ib)
        if not bool(eq.all().item()):
            # genuine 128-bit double collision (or a forced test): exact
            # host fallback keeps the answer right
            codes_all = _host_exact_codes(comb)
    if codes_all is None:
        boundary = torch.ones(n, dtype=torch.int64, device=comb.device)
        boundary[1:] = (~same).to(torch.int64)
        codes_sorted = torch.cumsum(boundary, 0) - 1
        codes_all = torch.empty(n, dtype=torch.int64, device=comb.device)
        codes_all[order] = codes_sorted
    out = []
    at = 0
    for c in cols:
        out.append(codes_all[at:at + len(c)])
        at += len(c)
    return out


def _expand_matches_gpu(bids: torch.Tensor, pids: torch.Tensor, ext):
    """HIP open-addressing hash join (ops/csrc/hash_join.hip): chain build +
    two-phase count/fill probe. Same contract as the sort-based matcher."""
    tkeys, theads, nxt = ext.hj_build_chain(bids)
    counts32 = ext.hj_probe_count(tkeys, theads, nxt, pids)
    counts = counts32.to(torch.int64)
    offsets = torch.cumsum(counts, 0) - counts
    total = int((offsets[-1] + counts[-1]).item()) if counts.numel() else 0
    probe_idx, build_idx = ext.hj_probe_fill(tkeys, theads, nxt, pids, offsets, total)
    return probe_idx, build_idx, counts
