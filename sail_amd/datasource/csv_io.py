"""CSV read/write (pyarrow host decode -> device upload).
ref: crates/sail-data-source/src/formats/csv (option surface)."""
from __future__ import annotations

import glob as _glob
import os
from typing import Dict, List, Tuple

import pyarrow as pa
import pyarrow.csv as pacsv

from ..engine import types as T
from .arrow_io import arrow_to_table, chunk_to_arrow
from .parquet_io import _engine_type


def _opts(options: Dict[str, str]):
    delim = options.get("sep", options.get("delimiter", ","))
    header = options.get("header", "true").lower() == "true"
    parse = pacsv.ParseOptions(delimiter=delim)
    read = pacsv.ReadOptions(autogenerate_column_names=not header)
    conv = pacsv.ConvertOptions(null_values=[options.get("nullValue", "")])
    return read, parse, conv


def _expand(paths: List[str]) -> List[str]:
    out = []
    for p in paths:
        if os.path.isdir(p):
            out.extend(sorted(_glob.glob(os.path.join(p, "*.csv"))))
        elif any(ch in p for ch in "*?["):
            out.extend(sorted(_glob.glob(p)))
        else:
            out.append(p)
    return out


def infer_schema(paths: List[str], options: Dict[str, str]) -> List[Tuple[str, T.DataType]]:
    files = _expand(paths)
    r, p, c = _opts(options)
    tbl = pacsv.read_csv(files[0], read_options=r, parse_options=p, convert_options=c)
    return [(f.name, _engine_type(f.type)) for f in tbl.schema]


def read(paths: List[str], schema, device, options: Dict[str, str]):
    files = _expand(paths)
    r, p, c = _opts(options)
    tables = [pacsv.read_csv(f, read_options=r, parse_options=p, convert_options=c)
              for f in files]
    tbl = pa.concat_tables(tables) if len(tables) > 1 else tables[0]
    return arrow_to_table(tbl, device=device)


def write(path: str, chunk, mode: str, options: Dict[str, str]):
    schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
    tbl = chunk_to_arrow(chunk, schema)
    if path.endswith(".csv"):
        parent = os.path.dirname(path)
        if parent:
            os.makedirs(parent, exist_ok=True)
        target = path
    else:
        os.makedirs(path, exist_ok=True)
        target = os.path.join(path, "part-00000.csv")
    if mode == "error" and os.path.exists(target):
        raise FileExistsError(target)
    pacsv.write_csv(tbl, target)
    return target
