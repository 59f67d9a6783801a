"""Parquet read/write.

Host decode via pyarrow (multi-threaded, column-pruned) -> device upload.
The GPU-native page decoder is a planned replacement for the decode stage
(ref: crates/sail-data-source/src/formats/parquet, SURVEY §2.9 scan path);
today the host path keeps the format surface complete.
"""
from __future__ import annotations

import glob as _glob
import os
from typing import Dict, List, Optional, Tuple

import pyarrow as pa
import pyarrow.parquet as pq

from ..engine import types as T
from .arrow_io import arrow_to_table, chunk_to_arrow


def _expand(paths: List[str]) -> List[str]:
    out = []
    for p in paths:
        if os.path.isdir(p):
            out.extend(sorted(_glob.glob(os.path.join(p, "**", "*.parquet"), recursive=True)))
        elif any(ch in p for ch in "*?["):
            out.extend(sorted(_glob.glob(p)))
        else:
            out.append(p)
    return out


def _engine_type(at: pa.DataType) -> T.DataType:
    if pa.types.is_decimal(at):
        return T.DecimalType(at.precision, at.scale)
    m = {"bool": T.BOOL, "int8": T.I8, "int16": T.I16, "int32": T.I32,
         "int64": T.I64, "float": T.F32, "double": T.F64, "date32[day]": T.DATE,
         "string": T.STRING, "large_string": T.STRING, "binary": T.BINARY}
    if pa.types.is_timestamp(at):
        return T.TIMESTAMP
    if pa.types.is_dictionary(at):
        return T.STRING
    s = str(at)
    if s in m:
        return m[s]
    raise ValueError(f"unsupported parquet type {at}")


def infer_schema(paths: List[str]) -> List[Tuple[str, T.DataType]]:
    files = _expand(paths)
    if not files:
        raise FileNotFoundError(f"no parquet files under {paths}")
    sch = pq.read_schema(files[0])
    return [(f.name, _engine_type(f.type)) for f in sch]


def read(paths: List[str], schema, device, options: Dict[str, str]):
    files = _expand(paths)
    cols = [n for n, _ in schema] if schema else None
    tbl = pq.read_table(files, columns=cols)
    return arrow_to_table(tbl, device=device)


def write(path: str, chunk, mode: str, options: Dict[str, str]):
    schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
    tbl = chunk_to_arrow(chunk, schema)
    if os.path.isdir(path) or path.endswith("/"):
        os.makedirs(path, exist_ok=True)
        target = os.path.join(path, "part-00000.parquet")
    else:
        parent = os.path.dirname(path)
        if parent:
            os.makedirs(parent, exist_ok=True)
        target = path if path.endswith(".parquet") else None
        if target is None:
            os.makedirs(path, exist_ok=True)
            target = os.path.join(path, "part-00000.parquet")
    if mode == "error" and os.path.exists(target):
        raise FileExistsError(target)
    compression = options.get("compression", "snappy")
    pq.write_table(tbl, target, compression=compression)
    return target
