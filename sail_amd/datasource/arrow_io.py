"""Arrow interop: Chunk <-> pyarrow (host boundary for clients/files)."""
from __future__ import annotations

import pyarrow as pa
import torch

from ..engine import types as T
from ..engine.chunk import Chunk
from ..engine.column import Column, StringColumn


def _arrow_type(t: T.DataType) -> pa.DataType:
    if isinstance(t, T.DecimalType):
        return pa.decimal128(max(t.precision, t.scale + 1), t.scale)
    m = {T.BooleanType: pa.bool_(), T.Int8Type: pa.int8(), T.Int16Type: pa.int16(),
         T.Int32Type: pa.int32(), T.Int64Type: pa.int64(), T.Float32Type: pa.float32(),
         T.Float64Type: pa.float64(), T.DateType: pa.date32(),
         T.TimestampType: pa.timestamp("us"), T.StringType: pa.string(),
         T.BinaryType: pa.binary(), T.NullType: pa.null()}
    return m[type(t)]


def column_to_arrow(c: Column, t: T.DataType) -> pa.Array:
    if isinstance(c, StringColumn):
        return pa.array(c.to_pylist(), type=pa.string())
    import numpy as np

    data = c.data.cpu().numpy()
    mask = None
    if c.validity is not None:
        mask = ~c.validity.cpu().numpy().astype(bool)
    if isinstance(t, T.DecimalType):
        from decimal import Decimal

        scale = t.scale
        vals = [None if (mask is not None and mask[i]) else Decimal(int(data[i])).scaleb(-scale)
                for i in range(len(data))]
        return pa.array(vals, type=_arrow_type(t))
    return pa.array(data, type=_arrow_type(t), mask=mask)


def chunk_to_arrow(chunk: Chunk, schema) -> pa.Table:
    arrays = []
    names = []
    for (n, t), c in zip(schema, chunk.columns):
        arrays.append(column_to_arrow(c, t))
        names.append(n)
    return pa.table(arrays, names=names)


def arrow_to_table(tbl: pa.Table, device="cpu", dict_encode=True):
    """pyarrow Table -> engine Table (host decode path)."""
    from ..engine.column import Table

    cols = {}
    for name, col in zip(tbl.column_names, tbl.columns):
        cols[name] = arrow_column(col, device, dict_encode)
    return Table(cols)


def arrow_column(col: pa.ChunkedArray, device="cpu", dict_encode=True) -> Column:
    import numpy as np

    at = col.type
    combined = col.combine_chunks() if isinstance(col, pa.ChunkedArray) else col
    if pa.types.is_dictionary(at):
        combined = combined.cast(at.value_type)
        at = at.value_type
    if pa.types.is_string(at) or pa.types.is_large_string(at):
        vals = combined.to_pylist()
        return StringColumn.from_pylist(vals, device=device, dict_encode=None if dict_encode else False)
    if pa.types.is_decimal(at):
        scale = at.scale
        vals = combined.to_pylist()
        data = torch.tensor([0 if v is None else int(v.scaleb(scale)) for v in vals],
                            dtype=torch.int64, device=device)
        validity = None
        if combined.null_count:
            validity = torch.tensor([v is not None for v in vals], dtype=torch.uint8, device=device)
        return Column(T.DecimalType(at.precision, scale), data, validity)
    npmap = {pa.bool_(): (torch.bool, T.BOOL), pa.int8(): (torch.int8, T.I8),
             pa.int16(): (torch.int16, T.I16), pa.int32(): (torch.int32, T.I32),
             pa.int64(): (torch.int64, T.I64), pa.float32(): (torch.float32, T.F32),
             pa.float64(): (torch.float64, T.F64), pa.date32(): (torch.int32, T.DATE)}
    if pa.types.is_timestamp(at):
        arr = combined.cast(pa.timestamp("us"))
        np_data = arr.to_numpy(zero_copy_only=False).astype("datetime64[us]").astype(np.int64)
        data = torch.from_numpy(np_data).to(device)
        validity = _validity(combined, device)
        return Column(T.TIMESTAMP, data, validity)
    for patype, (tt, et) in npmap.items():
        if at.equals(patype):
            np_data = combined.to_numpy(zero_copy_only=False)
            if pa.types.is_date32(at):
                np_data = np_data.astype("datetime64[D]").astype(np.int32)
            data = torch.from_numpy(np.ascontiguousarray(np_data)).to(tt).to(device)
            validity = _validity(combined, device)
            return Column(et, data, validity)
    raise ValueError(f"unsupported arrow type {at}")


def _validity(arr: pa.Array, device):
    if arr.null_count == 0:
        return None
    m = pa.compute.is_valid(arr).to_numpy(zero_copy_only=False)
    return torch.from_numpy(m.astype("uint8")).to(device)
