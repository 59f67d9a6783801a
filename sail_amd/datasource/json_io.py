"""NDJSON read (pyarrow host decode -> device upload).
ref: crates/sail-data-source/src/formats/json."""
from __future__ import annotations

import glob as _glob
import os
from typing import Dict, List, Tuple

import pyarrow as pa
import pyarrow.json as pajson

from ..engine import types as T
from .arrow_io import arrow_to_table
from .parquet_io import _engine_type


def _expand(paths: List[str]) -> List[str]:
    out = []
    for p in paths:
        if os.path.isdir(p):
            out.extend(sorted(_glob.glob(os.path.join(p, "*.json*"))))
        elif any(ch in p for ch in "*?["):
            out.extend(sorted(_glob.glob(p)))
        else:
            out.append(p)
    return out


def infer_schema(paths: List[str], options: Dict[str, str]) -> List[Tuple[str, T.DataType]]:
    files = _expand(paths)
    tbl = pajson.read_json(files[0])
    return [(f.name, _engine_type(f.type)) for f in tbl.schema]


def read(paths: List[str], schema, device, options: Dict[str, str]):
    files = _expand(paths)
    tables = [pajson.read_json(f) for f in files]
    tbl = pa.concat_tables(tables) if len(tables) > 1 else tables[0]
    return arrow_to_table(tbl, device=device)


def write(path: str, chunk, mode: str, options: Dict[str, str]):
    """NDJSON write (one JSON object per line, Spark json sink layout)."""
    import json as _json

    if path.endswith(".json"):
        parent = os.path.dirname(path)
        if parent:
            os.makedirs(parent, exist_ok=True)
        target = path
    else:
        os.makedirs(path, exist_ok=True)
        target = os.path.join(path, "part-00000.json")
    if mode == "error" and os.path.exists(target):
        raise FileExistsError(target)
    cols = [c.to_pylist() for c in chunk.columns]
    with open(target, "w") as f:
        for row in zip(*cols) if cols else []:
            f.write(_json.dumps({n: v for n, v in zip(chunk.names, row)},
                                default=str) + "\n")
    return target
