"""Columnar device data model.

A Column is a torch tensor living in HBM (or host RAM for the CPU path) plus
optional validity and string payloads. Unlike the reference — which streams
8192-row Arrow batches through pull-based operator trees
(ref: crates/sail-common/src/config/application.yaml execution.batch_size) —
this engine holds *whole table partitions* resident in the 288 GB HBM of each
GPU and runs operators over entire columns in one kernel launch.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from . import types as T


def _dev(device) -> torch.device:
    return torch.device(device) if not isinstance(device, torch.device) else device


class Column:
    """A typed column of values.

    Fixed-width types: `data` holds the values (decimal -> scaled int64,
    date -> int32 days, timestamp -> int64 micros).
    Strings: `data` holds int64 offsets? No — see StringColumn.

    `validity`: optional uint8 mask tensor, 1 = valid. None means all-valid.
    """

    __slots__ = ("dtype", "data", "validity")

    def __init__(self, dtype: T.DataType, data: torch.Tensor, validity: Optional[torch.Tensor] = None):
        self.dtype = dtype
        self.data = data
        self.validity = validity

    # -- construction ------------------------------------------------------
    @staticmethod
    def from_values(values: Sequence, dtype: T.DataType, device="cpu") -> "Column":
        device = _dev(device)
        if isinstance(dtype, T.StringType):
            return StringColumn.from_pylist(list(values), device=device, dtype=dtype)
        if isinstance(dtype, T.ArrayType):  # nested lists
            return ListColumn.from_pylist(list(values), dtype.element,
                                          device=device)
        if isinstance(dtype, T.MapType):  # rows are dicts (or None)
            return MapColumn.from_pylist(list(values), dtype.key,
                                         dtype.value, device=device)
        if isinstance(dtype, T.StructType):  # rows are dicts (or None)
            vals = list(values)
            fields = []
            for f in dtype.fields:
                fv = [None if r is None else r.get(f.name) for r in vals]
                fields.append((f.name,
                               Column.from_values(fv, f.dtype, device=device)))
            validity = None
            if any(r is None for r in vals):
                validity = torch.tensor(
                    [0 if r is None else 1 for r in vals],
                    dtype=torch.uint8, device=device)
            return StructColumn(fields, validity, dtype=dtype)
        validity = None
        if any(v is None for v in values):
            validity = torch.tensor([0 if v is None else 1 for v in values], dtype=torch.uint8, device=device)
        if isinstance(dtype, T.DecimalType):
            scale = 10 ** dtype.scale
            vals = [0 if v is None else int(round(float(v) * scale)) for v in values]
            data = torch.tensor(vals, dtype=torch.int64, device=device)
        elif isinstance(dtype, T.DateType):
            vals = [0 if v is None else _to_days(v) for v in values]
            data = torch.tensor(vals, dtype=torch.int32, device=device)
        elif isinstance(dtype, T.BooleanType):
            data = torch.tensor([bool(v) for v in [False if v is None else v for v in values]], dtype=torch.bool, device=device)
        else:
            zero = 0 if dtype.is_integer or dtype.is_temporal else 0.0
            vals = [zero if v is None else v for v in values]
            data = torch.tensor(vals, dtype=dtype.storage, device=device)
        return Column(dtype, data, validity)

    # -- basics ------------------------------------------------------------
    def __len__(self) -> int:
        return int(self.data.shape[0])

    @property
    def device(self) -> torch.device:
        return self.data.device

    @property
    def is_cuda(self) -> bool:
        return self.data.is_cuda

    def to(self, device) -> "Column":
        device = _dev(device)
        if self.device == device:
            return self
        return Column(self.dtype, self.data.to(device),
                      self.validity.to(device) if self.validity is not None else None)

    def gather(self, indices: torch.Tensor) -> "Column":
        v = self.validity.index_select(0, indices) if self.validity is not None else None
        return Column(self.dtype, self.data.index_select(0, indices), v)

    def filter(self, mask: torch.Tensor) -> "Column":
        v = self.validity[mask] if self.validity is not None else None
        return Column(self.dtype, self.data[mask], v)

    def slice(self, start: int, length: int) -> "Column":
        v = self.validity[start : start + length] if self.validity is not None else None
        return Column(self.dtype, self.data[start : start + length], v)

    def null_count(self) -> int:
        if self.validity is None:
            return 0
        return int(len(self) - int(self.validity.sum().item()))

    def valid_mask(self) -> torch.Tensor:
        if self.validity is None:
            return torch.ones(len(self), dtype=torch.bool, device=self.device)
        return self.validity.to(torch.bool)

    # -- host conversion ---------------------------------------------------
    def to_pylist(self) -> List:
        vals = self.data.cpu()
        out: List = []
        scale = 10 ** self.dtype.scale if isinstance(self.dtype, T.DecimalType) else None
        vmask = self.validity.cpu().tolist() if self.validity is not None else None
        lst = vals.tolist()
        for i, v in enumerate(lst):
            if vmask is not None and not vmask[i]:
                out.append(None)
            elif scale is not None:
                out.append(v / scale)
            elif isinstance(self.dtype, T.DateType):
                out.append(_from_days(v))
            elif isinstance(self.dtype, T.TimeType):
                out.append(_from_time_us(v))
            else:
                out.append(v)
        return out

    def __repr__(self):
        return f"Column({self.dtype!r}, n={len(self)}, dev={self.device}, nulls={self.null_count()})"


class StringColumn(Column):
    """Arrow-style UTF-8 column: int64 offsets (n+1) + uint8 bytes, with an
    optional dictionary encoding (codes int32 into a unique-values column).

    When `codes` is not None the column is dictionary-encoded: `offsets`/
    `bytes_` describe the *dictionary* values and `codes[i]` picks row i's
    value (-1 = null). Low-cardinality TPC-H columns (flags, status,
    segments, ...) stay dict-encoded end-to-end so that predicates and joins
    on them are integer ops on device.
    """

    __slots__ = ("offsets", "bytes_", "codes")

    def __init__(self, offsets: torch.Tensor, bytes_: torch.Tensor,
                 validity: Optional[torch.Tensor] = None,
                 codes: Optional[torch.Tensor] = None,
                 dtype: Optional[T.DataType] = None):
        n = (codes.shape[0] if codes is not None else offsets.shape[0] - 1)
        # `data` for a string column is a row-count-sized placeholder view used
        # only for len()/device; real payloads are offsets/bytes_/codes.
        anchor = codes if codes is not None else offsets[:-1] if offsets.numel() else offsets
        super().__init__(dtype or T.STRING, anchor, validity)
        self.offsets = offsets
        self.bytes_ = bytes_
        self.codes = codes

    # -- construction ------------------------------------------------------
    @staticmethod
    def from_pylist(values: List[Optional[str]], device="cpu", dict_encode: Optional[bool] = None,
                    dtype: Optional[T.DataType] = None) -> "StringColumn":
        device = _dev(device)
        n = len(values)
        validity = None
        if any(v is None for v in values):
            validity = torch.tensor([0 if v is None else 1 for v in values], dtype=torch.uint8, device=device)
        empty = b"" if isinstance(dtype, T.BinaryType) else ""
        values = [empty if v is None else v for v in values]
        uniq = set(v for v in values if v is not None)
        if dict_encode is None:
            dict_encode = n > 64 and len(uniq) * 16 < n
        if dict_encode:
            udict = sorted(uniq)
            idx = {s: i for i, s in enumerate(udict)}
            codes = torch.tensor([(-1 if v is None else idx[v]) for v in values], dtype=torch.int32, device=device)
            offsets, bytes_ = _pack_strings(udict, device)
            return StringColumn(offsets, bytes_, validity, codes, dtype=dtype)
        offsets, bytes_ = _pack_strings(values, device)
        return StringColumn(offsets, bytes_, validity, None, dtype=dtype)

    @staticmethod
    def from_buffers(offsets: torch.Tensor, bytes_: torch.Tensor, validity=None, codes=None) -> "StringColumn":
        return StringColumn(offsets, bytes_, validity, codes)

    # -- basics ------------------------------------------------------------
    def __len__(self) -> int:
        if self.codes is not None:
            return int(self.codes.shape[0])
        return int(self.offsets.shape[0]) - 1

    @property
    def is_dict(self) -> bool:
        return self.codes is not None

    @property
    def dict_size(self) -> int:
        return int(self.offsets.shape[0]) - 1

    def to(self, device) -> "StringColumn":
        device = _dev(device)
        if self.device == device:
            return self
        return StringColumn(self.offsets.to(device), self.bytes_.to(device),
                            self.validity.to(device) if self.validity is not None else None,
                            self.codes.to(device) if self.codes is not None else None,
                            dtype=self.dtype)

    @property
    def device(self) -> torch.device:
        return self.offsets.device if self.codes is None else self.codes.device

    def gather(self, indices: torch.Tensor) -> "StringColumn":
        v = self.validity.index_select(0, indices) if self.validity is not None else None
        if self.codes is not None:
            return StringColumn(self.offsets, self.bytes_, v, self.codes.index_select(0, indices), dtype=self.dtype)
        # materialized gather of raw strings
        offs, byts = _gather_strings(self.offsets, self.bytes_, indices)
        return StringColumn(offs, byts, v, None, dtype=self.dtype)

    def filter(self, mask: torch.Tensor) -> "StringColumn":
        return self.gather(torch.nonzero(mask, as_tuple=False).squeeze(1))

    def slice(self, start: int, length: int) -> "StringColumn":
        idx = torch.arange(start, start + length, device=self.device)
        return self.gather(idx)

    def decode_dict(self) -> "StringColumn":
        """Materialize a dict-encoded column into raw offsets/bytes."""
        if self.codes is None:
            return self
        offs, byts = _gather_strings(self.offsets, self.bytes_, self.codes.clamp_min(0).to(torch.int64))
        return StringColumn(offs, byts, self.validity, None, dtype=self.dtype)

    def dict_code_of(self, value: str) -> int:
        """Code of `value` in the (sorted) dictionary, -1 if absent.
        Binary search decoding O(log n) entries — never the whole
        dictionary (a 1M-entry URL dictionary costs ~150 ms to decode;
        ClickBench q36 hit this on every `URL <> ''`)."""
        nd = self.offsets.shape[0] - 1
        if nd <= 0:
            return -1
        offs = self.offsets.cpu()
        target = value.encode("utf-8")
        byts = None
        lo, hi = 0, nd - 1
        while lo <= hi:
            mid = (lo + hi) // 2
            b0, b1 = int(offs[mid]), int(offs[mid + 1])
            entry = bytes(self.bytes_[b0:b1].cpu().numpy())
            if entry == target:
                return mid
            if entry < target:
                lo = mid + 1
            else:
                hi = mid - 1
        return -1

    def dict_values(self) -> List[str]:
        """Host copy of the dictionary (or all values if not dict-encoded).
        BINARY-typed columns return raw bytes (no utf-8 decode)."""
        offs = self.offsets.cpu().numpy()
        byts = self.bytes_.cpu().numpy().tobytes()
        if isinstance(self.dtype, T.BinaryType):
            return [byts[offs[i]:offs[i + 1]] for i in range(len(offs) - 1)]
        return [byts[offs[i]:offs[i + 1]].decode("utf-8", "replace") for i in range(len(offs) - 1)]

    def to_pylist(self) -> List[Optional[str]]:
        vals = self.dict_values()
        vmask = self.validity.cpu().tolist() if self.validity is not None else None
        if self.codes is not None:
            codes = self.codes.cpu().tolist()
            out = [vals[c] if c >= 0 else None for c in codes]
        else:
            out = list(vals)
        if vmask is not None:
            out = [None if not vmask[i] else v for i, v in enumerate(out)]
        return out

    def __repr__(self):
        kind = f"dict[{self.dict_size}]" if self.is_dict else "utf8"
        return f"StringColumn({kind}, n={len(self)}, dev={self.device})"


def _pack_strings(strs: List[str], device) -> tuple:
    enc = [s if isinstance(s, (bytes, bytearray)) else s.encode("utf-8")
           for s in strs]
    lens = np.fromiter((len(e) for e in enc), dtype=np.int64, count=len(enc))
    offsets = np.zeros(len(enc) + 1, dtype=np.int64)
    np.cumsum(lens, out=offsets[1:])
    data = b"".join(enc)
    byts = torch.frombuffer(bytearray(data), dtype=torch.uint8) if data else torch.zeros(0, dtype=torch.uint8)
    return torch.from_numpy(offsets).to(device), byts.to(device)


def _gather_strings(offsets: torch.Tensor, bytes_: torch.Tensor, indices: torch.Tensor,
                    byte_chunk: int = 1 << 28):
    """Gather rows of a raw string column by index (torch path).

    Output is assembled in ~256 MB byte slices: the per-byte int64 index
    temporaries are 24 B per output byte and must stay bounded."""
    indices = indices.to(torch.int64)
    starts = offsets.index_select(0, indices)
    ends = offsets.index_select(0, indices + 1)
    lens = ends - starts
    n = indices.shape[0]
    dev = offsets.device
    out_offsets = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=out_offsets[1:])
    total = int(out_offsets[-1].item())
    if total == 0:
        return out_offsets, torch.zeros(0, dtype=torch.uint8, device=bytes_.device)
    if total <= byte_chunk:
        pos = torch.arange(total, device=dev)
        row = torch.searchsorted(out_offsets[1:], pos, right=True)
        src = starts.index_select(0, row) + (pos - out_offsets.index_select(0, row))
        return out_offsets, bytes_.index_select(0, src)
    # row-sliced assembly
    parts = []
    row_start = 0
    while row_start < n:
        # find the row range covering ~byte_chunk bytes
        target = int(out_offsets[row_start].item()) + byte_chunk
        row_end = int(torch.searchsorted(out_offsets, torch.tensor(target, device=dev)).item())
        row_end = max(row_start + 1, min(row_end, n))
        seg_lo = int(out_offsets[row_start].item())
        seg_hi = int(out_offsets[row_end].item())
        seg = seg_hi - seg_lo
        pos = torch.arange(seg, device=dev)
        local_offs = out_offsets[row_start : row_end + 1] - seg_lo
        row = torch.searchsorted(local_offs[1:], pos, right=True)
        src = starts[row_start:row_end].index_select(0, row) + (pos - local_offs.index_select(0, row))
        parts.append(bytes_.index_select(0, src))
        row_start = row_end
    return out_offsets, torch.cat(parts)


def _from_time_us(us: int):
    import datetime as _dt2

    us = int(us)
    return _dt2.time((us // 3_600_000_000) % 24,
                     (us // 60_000_000) % 60,
                     (us // 1_000_000) % 60, us % 1_000_000)


def _to_days(v) -> int:
    import datetime as _dt

    if isinstance(v, int):
        return v
    if isinstance(v, str):
        y, m, d = v.split("-")
        v = _dt.date(int(y), int(m), int(d))
    if isinstance(v, _dt.datetime):
        v = v.date()
    return (v - _dt.date(1970, 1, 1)).days


def _from_days(days: int):
    import datetime as _dt

    return _dt.date(1970, 1, 1) + _dt.timedelta(days=int(days))


@dataclass
class Table:
    """A named collection of equal-length columns (one partition)."""

    columns: Dict[str, Column] = field(default_factory=dict)

    @property
    def num_rows(self) -> int:
        for c in self.columns.values():
            return len(c)
        return 0

    @property
    def names(self) -> List[str]:
        return list(self.columns.keys())

    @property
    def device(self) -> torch.device:
        for c in self.columns.values():
            return c.device
        return torch.device("cpu")

    def column(self, name: str) -> Column:
        return self.columns[name]

    def to(self, device) -> "Table":
        return Table({k: v.to(device) for k, v in self.columns.items()})

    def gather(self, indices: torch.Tensor) -> "Table":
        return Table({k: v.gather(indices) for k, v in self.columns.items()})

    def filter(self, mask: torch.Tensor) -> "Table":
        idx = torch.nonzero(mask, as_tuple=False).squeeze(1)
        return self.gather(idx)

    def select(self, names: Sequence[str]) -> "Table":
        return Table({n: self.columns[n] for n in names})

    def with_column(self, name: str, col: Column) -> "Table":
        cols = dict(self.columns)
        cols[name] = col
        return Table(cols)

    def rename(self, mapping: Dict[str, str]) -> "Table":
        return Table({mapping.get(k, k): v for k, v in self.columns.items()})

    def to_pydict(self) -> Dict[str, List]:
        return {k: v.to_pylist() for k, v in self.columns.items()}

    def to_rows(self) -> List[tuple]:
        d = self.to_pydict()
        names = list(d.keys())
        return [tuple(d[n][i] for n in names) for i in range(self.num_rows)]

    @staticmethod
    def from_pydict(data: Dict[str, Sequence], schema: Dict[str, T.DataType], device="cpu") -> "Table":
        return Table({k: Column.from_values(data[k], schema[k], device=device) for k in schema})

    def __repr__(self):
        return f"Table(rows={self.num_rows}, cols={self.names})"


class ListColumn(Column):
    """Arrow-style list column: int64 offsets (n+1) into a child element
    column (fixed-width, string, or nested list). Array operators are
    segment ops over the flat child — the layout GPU kernels and
    repeat_interleave/cumsum both like (ref: Spark ArrayType semantics;
    implementation is torch segment ops, not a port)."""

    __slots__ = ("offsets", "child")

    def __init__(self, offsets: torch.Tensor, child: Column,
                 validity: Optional[torch.Tensor] = None,
                 dtype: Optional[T.DataType] = None):
        anchor = offsets[:-1] if offsets.numel() else offsets
        super().__init__(dtype or T.ArrayType(child.dtype), anchor, validity)
        self.offsets = offsets
        self.child = child

    @staticmethod
    def from_pylist(values: List[Optional[list]], elem_type: T.DataType,
                    device="cpu") -> "ListColumn":
        device = _dev(device)
        validity = None
        if any(v is None for v in values):
            validity = torch.tensor([0 if v is None else 1 for v in values],
                                    dtype=torch.uint8, device=device)
        lens = [0 if v is None else len(v) for v in values]
        offsets = torch.zeros(len(values) + 1, dtype=torch.int64, device=device)
        if values:
            torch.cumsum(torch.tensor(lens, dtype=torch.int64, device=device),
                         0, out=offsets[1:])
        flat: List = []
        for v in values:
            if v is not None:
                flat.extend(v)
        child = Column.from_values(flat, elem_type, device=device)
        return ListColumn(offsets, child, validity)

    def __len__(self) -> int:
        return int(self.offsets.shape[0]) - 1

    @property
    def device(self) -> torch.device:
        return self.offsets.device

    def lengths(self) -> torch.Tensor:
        return self.offsets[1:] - self.offsets[:-1]

    def segment_ids(self) -> torch.Tensor:
        """child row -> parent row map."""
        n = len(self)
        return torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=self.device), self.lengths())

    def to(self, device) -> "ListColumn":
        device = _dev(device)
        if self.device == device:
            return self
        return ListColumn(self.offsets.to(device), self.child.to(device),
                          self.validity.to(device) if self.validity is not None else None,
                          self.dtype)

    def gather(self, indices: torch.Tensor) -> "ListColumn":
        lens = self.lengths().index_select(0, indices)
        new_off = torch.zeros(indices.shape[0] + 1, dtype=torch.int64, device=self.device)
        torch.cumsum(lens, 0, out=new_off[1:])
        total = int(new_off[-1].item()) if indices.numel() else 0
        starts = self.offsets[:-1].index_select(0, indices)
        # child index = start_of_selected_row + position inside that row
        pos = torch.arange(total, dtype=torch.int64, device=self.device) \
            - torch.repeat_interleave(new_off[:-1], lens)
        child_idx = torch.repeat_interleave(starts, lens) + pos
        v = self.validity.index_select(0, indices) if self.validity is not None else None
        return ListColumn(new_off, self.child.gather(child_idx), v, self.dtype)

    def filter(self, mask: torch.Tensor) -> "ListColumn":
        return self.gather(torch.nonzero(mask, as_tuple=False).flatten())

    def slice(self, start: int, length: int) -> "ListColumn":
        off = self.offsets[start : start + length + 1]
        base = int(off[0].item()) if off.numel() else 0
        child = self.child.slice(base, int(off[-1].item()) - base if off.numel() else 0)
        v = self.validity[start : start + length] if self.validity is not None else None
        return ListColumn(off - base, child, v, self.dtype)

    def null_count(self) -> int:
        if self.validity is None:
            return 0
        return int(len(self) - int(self.validity.sum().item()))

    def valid_mask(self) -> torch.Tensor:
        if self.validity is None:
            return torch.ones(len(self), dtype=torch.bool, device=self.device)
        return self.validity.to(torch.bool)

    def to_pylist(self) -> List:
        flat = self.child.to_pylist()
        offs = self.offsets.cpu().tolist()
        vmask = self.validity.cpu().tolist() if self.validity is not None else None
        out: List = []
        for i in range(len(self)):
            if vmask is not None and not vmask[i]:
                out.append(None)
            else:
                out.append(flat[offs[i] : offs[i + 1]])
        return out

    def __repr__(self):
        return f"ListColumn({self.dtype!r}, n={len(self)}, dev={self.device})"


class MapColumn(Column):
    """Map column: list-of-entries layout — int64 offsets (n+1) into parallel
    keys/values child columns (Arrow map layout; ref: Spark MapType)."""

    __slots__ = ("offsets", "keys", "values")

    def __init__(self, offsets: torch.Tensor, keys: Column, values: Column,
                 validity: Optional[torch.Tensor] = None,
                 dtype: Optional[T.DataType] = None):
        anchor = offsets[:-1] if offsets.numel() else offsets
        super().__init__(dtype or T.MapType(keys.dtype, values.dtype),
                         anchor, validity)
        self.offsets = offsets
        self.keys = keys
        self.values = values

    @staticmethod
    def from_pylist(values: List[Optional[dict]], key_t: T.DataType,
                    val_t: T.DataType, device="cpu") -> "MapColumn":
        device = _dev(device)
        validity = None
        if any(v is None for v in values):
            validity = torch.tensor([0 if v is None else 1 for v in values],
                                    dtype=torch.uint8, device=device)
        lens = [0 if v is None else len(v) for v in values]
        offsets = torch.zeros(len(values) + 1, dtype=torch.int64, device=device)
        if values:
            torch.cumsum(torch.tensor(lens, dtype=torch.int64, device=device),
                         0, out=offsets[1:])
        fk: List = []
        fv: List = []
        for v in values:
            if v is not None:
                fk.extend(v.keys())
                fv.extend(v.values())
        return MapColumn(offsets, Column.from_values(fk, key_t, device=device),
                         Column.from_values(fv, val_t, device=device), validity)

    def __len__(self) -> int:
        return int(self.offsets.shape[0]) - 1

    @property
    def device(self) -> torch.device:
        return self.offsets.device

    def lengths(self) -> torch.Tensor:
        return self.offsets[1:] - self.offsets[:-1]

    def segment_ids(self) -> torch.Tensor:
        n = len(self)
        return torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=self.device), self.lengths())

    def _as_lists(self):
        """View keys/values as ListColumns sharing this map's offsets."""
        return (ListColumn(self.offsets, self.keys, self.validity),
                ListColumn(self.offsets, self.values, self.validity))

    def to(self, device) -> "MapColumn":
        device = _dev(device)
        if self.device == device:
            return self
        return MapColumn(self.offsets.to(device), self.keys.to(device),
                         self.values.to(device),
                         self.validity.to(device) if self.validity is not None else None,
                         self.dtype)

    def gather(self, indices: torch.Tensor) -> "MapColumn":
        kl, vl = self._as_lists()
        gk = kl.gather(indices)
        gv = vl.gather(indices)
        return MapColumn(gk.offsets, gk.child, gv.child, gk.validity, self.dtype)

    def filter(self, mask: torch.Tensor) -> "MapColumn":
        return self.gather(torch.nonzero(mask, as_tuple=False).flatten())

    def slice(self, start: int, length: int) -> "MapColumn":
        kl, vl = self._as_lists()
        sk = kl.slice(start, length)
        sv = vl.slice(start, length)
        return MapColumn(sk.offsets, sk.child, sv.child, sk.validity, self.dtype)

    def null_count(self) -> int:
        if self.validity is None:
            return 0
        return int(len(self) - int(self.validity.sum().item()))

    def valid_mask(self) -> torch.Tensor:
        if self.validity is None:
            return torch.ones(len(self), dtype=torch.bool, device=self.device)
        return self.validity.to(torch.bool)

    def to_pylist(self) -> List:
        ks = self.keys.to_pylist()
        vs = self.values.to_pylist()
        offs = self.offsets.cpu().tolist()
        vmask = self.validity.cpu().tolist() if self.validity is not None else None
        out: List = []
        for i in range(len(self)):
            if vmask is not None and not vmask[i]:
                out.append(None)
            else:
                out.append(dict(zip(ks[offs[i]:offs[i + 1]], vs[offs[i]:offs[i + 1]])))
        return out

    def __repr__(self):
        return f"MapColumn({self.dtype!r}, n={len(self)}, dev={self.device})"


class StructColumn(Column):
    """Struct column: parallel named children (Arrow struct layout)."""

    __slots__ = ("children_",)

    def __init__(self, names_children, validity: Optional[torch.Tensor] = None,
                 dtype: Optional[T.DataType] = None):
        self.children_ = list(names_children)  # [(name, Column)]
        n = len(self.children_[0][1]) if self.children_ else 0
        anchor = (self.children_[0][1].data if self.children_
                  else torch.zeros(0, dtype=torch.int64))
        st = dtype or T.StructType(tuple(
            T.StructField(nm, c.dtype) for nm, c in self.children_))
        super().__init__(st, anchor, validity)

    def __len__(self) -> int:
        return len(self.children_[0][1]) if self.children_ else 0

    @property
    def device(self) -> torch.device:
        return self.children_[0][1].device if self.children_ else torch.device("cpu")

    def field(self, name: str) -> Column:
        for nm, c in self.children_:
            if nm.lower() == name.lower():
                return c
        raise KeyError(name)

    def to(self, device) -> "StructColumn":
        device = _dev(device)
        if self.device == device:
            return self
        return StructColumn([(nm, c.to(device)) for nm, c in self.children_],
                            self.validity.to(device) if self.validity is not None else None,
                            self.dtype)

    def gather(self, indices: torch.Tensor) -> "StructColumn":
        v = self.validity.index_select(0, indices) if self.validity is not None else None
        return StructColumn([(nm, c.gather(indices)) for nm, c in self.children_],
                            v, self.dtype)

    def filter(self, mask: torch.Tensor) -> "StructColumn":
        return self.gather(torch.nonzero(mask, as_tuple=False).flatten())

    def slice(self, start: int, length: int) -> "StructColumn":
        v = self.validity[start:start + length] if self.validity is not None else None
        return StructColumn([(nm, c.slice(start, length)) for nm, c in self.children_],
                            v, self.dtype)

    def null_count(self) -> int:
        if self.validity is None:
            return 0
        return int(len(self) - int(self.validity.sum().item()))

    def valid_mask(self) -> torch.Tensor:
        if self.validity is None:
            return torch.ones(len(self), dtype=torch.bool, device=self.device)
        return self.validity.to(torch.bool)

    def to_pylist(self) -> List:
        lists = [(nm, c.to_pylist()) for nm, c in self.children_]
        vmask = self.validity.cpu().tolist() if self.validity is not None else None
        out: List = []
        for i in range(len(self)):
            if vmask is not None and not vmask[i]:
                out.append(None)
            else:
                out.append({nm: vals[i] for nm, vals in lists})
        return out

    def __repr__(self):
        return f"StructColumn({self.dtype!r}, n={len(self)})"
