"""Data source registry (parquet/csv/json/delta + user-defined formats).

ref: crates/sail-data-source/src/formats/ for the reference's format set;
user-defined Python data sources mirror formats/python/ (discovery via an
in-process registry instead of entry points — no package installation in
the image).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..engine import types as T

#: user-defined formats: name -> object with infer_schema(paths, options),
#: read(paths, schema, device, options) -> Table, optional
#: write(path, chunk, mode, options)
_USER_FORMATS: Dict[str, object] = {}


def register_format(name: str, source) -> None:
    """Register a Python data source (ref: sail-data-source
    formats/python/ — user-defined DataSource classes). `source` provides:
      infer_schema(paths, options) -> [(name, DataType)]
      read(paths, schema, device, options) -> engine.column.Table
      write(path, chunk, mode, options)     (optional)
    """
    _USER_FORMATS[name.lower()] = source


def infer_source_schema(fmt: str, paths: List[str], options: Dict[str, str]):
    fmt = fmt.lower()
    from . import parquet_io, csv_io

    if fmt == "parquet":
        return parquet_io.infer_schema(paths)
    if fmt == "csv":
        return csv_io.infer_schema(paths, options)
    if fmt == "json":
        from . import json_io

        return json_io.infer_schema(paths, options)
    if fmt == "delta":
        from . import delta

        return delta.infer_schema(paths, options)
    if fmt == "iceberg":
        from . import iceberg

        return iceberg.infer_schema(paths, options)
    if fmt == "text":
        from . import text_io

        return text_io.text_infer_schema(paths, options)
    if fmt in ("binary", "binaryfile"):
        from . import text_io

        return text_io.binary_infer_schema(paths, options)
    if fmt == "arrow":
        from . import text_io

        return text_io.arrow_infer_schema(paths, options)
    if fmt in _USER_FORMATS:
        return _USER_FORMATS[fmt].infer_schema(paths, options)
    raise ValueError(f"unsupported format {fmt}")


def read_source(fmt: str, paths: List[str], options: Dict[str, str], schema, device):
    fmt = fmt.lower()
    from . import parquet_io, csv_io

    if fmt == "parquet":
        return parquet_io.read(paths, schema, device, options)
    if fmt == "csv":
        return csv_io.read(paths, schema, device, options)
    if fmt == "json":
        from . import json_io

        return json_io.read(paths, schema, device, options)
    if fmt == "delta":
        from . import delta

        return delta.read(paths, schema, device, options)
    if fmt == "iceberg":
        from . import iceberg

        return iceberg.read(paths, schema, device, options)
    if fmt == "text":
        from . import text_io

        return text_io.text_read(paths, schema, device, options)
    if fmt in ("binary", "binaryfile"):
        from . import text_io

        return text_io.binary_read(paths, schema, device, options)
    if fmt == "arrow":
        from . import text_io

        return text_io.arrow_read(paths, schema, device, options)
    if fmt in _USER_FORMATS:
        return _USER_FORMATS[fmt].read(paths, schema, device, options)
    raise ValueError(f"unsupported format {fmt}")


def write_source(fmt: str, path: str, chunk, mode: str, options, partition_by):
    fmt = fmt.lower()
    if partition_by and fmt == "parquet":
        return _write_partitioned(path, chunk, mode, options, partition_by)
    if partition_by and fmt == "iceberg":
        from . import iceberg

        return iceberg.write_partitioned(path, chunk, mode, options,
                                         partition_by)
    from . import parquet_io, csv_io

    if fmt == "parquet":
        return parquet_io.write(path, chunk, mode, options)
    if fmt == "csv":
        return csv_io.write(path, chunk, mode, options)
    if fmt == "json":
        from . import json_io

        return json_io.write(path, chunk, mode, options)
    if fmt == "delta":
        from . import delta

        return delta.write(path, chunk, mode, options)
    if fmt == "iceberg":
        from . import iceberg

        return iceberg.write(path, chunk, mode, options)
    if fmt == "text":
        from . import text_io

        return text_io.text_write(path, chunk, mode, options)
    if fmt == "arrow":
        from . import text_io

        return text_io.arrow_write(path, chunk, mode, options)
    if fmt in _USER_FORMATS and hasattr(_USER_FORMATS[fmt], "write"):
        return _USER_FORMATS[fmt].write(path, chunk, mode, options)
    raise ValueError(f"unsupported write format {fmt}")


def _write_partitioned(path, chunk, mode, options, partition_by):
    """Hive-style partitioned parquet write: one `key=value/.../part.parquet`
    per distinct partition tuple (ref: sail-data-source listing sink
    planning)."""
    import os

    import torch

    from . import parquet_io
    from ..engine.aggregates import group_ids
    from ..engine.chunk import Chunk as _Chunk

    names = [n.lower() for n in chunk.names]
    pidx = []
    for k in partition_by:
        if k.lower() not in names:
            raise ValueError(f"partitionBy column {k} not in output")
        pidx.append(names.index(k.lower()))
    key_cols = [chunk.columns[i] for i in pidx]
    gid, rep, ng = group_ids(key_cols)
    data_idx = [i for i in range(len(chunk.columns)) if i not in pidx]
    written = []
    for g in range(ng):
        rows = torch.nonzero(gid == g, as_tuple=False).flatten()
        sub = _Chunk([chunk.columns[i].gather(rows) for i in data_idx],
                     [chunk.names[i] for i in data_idx])
        vals = [key_cols[j].gather(rep[g:g + 1]).to_pylist()[0]
                for j in range(len(key_cols))]
        sub_dir = os.path.join(path, *[f"{k}={v}" for k, v in
                                       zip(partition_by, vals)])
        os.makedirs(sub_dir, exist_ok=True)
        written.append(parquet_io.write(sub_dir, sub, mode, options))
    return written
