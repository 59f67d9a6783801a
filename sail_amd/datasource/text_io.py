"""text / binaryFile / arrow formats (ref: sail-data-source formats/
{text,binary,arrow})."""
from __future__ import annotations

import glob as _glob
import os
from typing import Dict, List

from ..engine import types as T
from ..engine.column import Column, StringColumn, Table


def _expand(paths: List[str], ext: str = "") -> List[str]:
    out = []
    for p in paths:
        if os.path.isdir(p):
            out.extend(sorted(_glob.glob(os.path.join(p, "**", f"*{ext}"),
                                         recursive=True)))
        elif any(ch in p for ch in "*?["):
            out.extend(sorted(_glob.glob(p)))
        else:
            out.append(p)
    return [f for f in out if os.path.isfile(f)
            and not os.path.basename(f).startswith((".", "_"))]


# -- text: one row per line --------------------------------------------------

def text_infer_schema(paths, options):
    return [("value", T.STRING)]


def text_read(paths, schema, device, options):
    lines: List[str] = []
    for f in _expand(paths):
        with open(f, "r", errors="replace") as fh:
            lines.extend(ln.rstrip("\n") for ln in fh)
    return Table({"value": StringColumn.from_pylist(lines, device=device,
                                                    dict_encode=False)})


def text_write(path, chunk, mode, options):
    os.makedirs(path, exist_ok=True)
    target = os.path.join(path, "part-00000.txt")
    vals = chunk.columns[0].to_pylist()
    with open(target, "w") as fh:
        for v in vals:
            fh.write(("" if v is None else str(v)) + "\n")
    return target


# -- binaryFile: whole file per row ------------------------------------------

def binary_infer_schema(paths, options):
    return [("path", T.STRING), ("length", T.I64), ("content", T.BINARY)]


def binary_read(paths, schema, device, options):
    names, lens, blobs = [], [], []
    for f in _expand(paths):
        with open(f, "rb") as fh:
            b = fh.read()
        names.append(f)
        lens.append(len(b))
        blobs.append(b.decode("latin-1"))  # byte-preserving string storage
    return Table({
        "path": StringColumn.from_pylist(names, device=device, dict_encode=False),
        "length": Column.from_values(lens, T.I64, device=device),
        "content": StringColumn.from_pylist(blobs, device=device,
                                            dict_encode=False, dtype=T.BINARY),
    })


# -- arrow: IPC files --------------------------------------------------------

def arrow_infer_schema(paths, options):
    import pyarrow.ipc as ipc

    from .parquet_io import _engine_type

    files = _expand(paths, ".arrow") or _expand(paths)
    with ipc.open_file(files[0]) as r:
        return [(f.name, _engine_type(f.type)) for f in r.schema]


def arrow_read(paths, schema, device, options):
    import pyarrow as pa
    import pyarrow.ipc as ipc

    from .arrow_io import arrow_to_table

    files = _expand(paths, ".arrow") or _expand(paths)
    tables = []
    for f in files:
        with ipc.open_file(f) as r:
            tables.append(r.read_all())
    return arrow_to_table(pa.concat_tables(tables), device=device)


def arrow_write(path, chunk, mode, options):
    import pyarrow.ipc as ipc

    from .arrow_io import chunk_to_arrow

    os.makedirs(path, exist_ok=True)
    target = os.path.join(path, "part-00000.arrow")
    tbl = chunk_to_arrow(chunk, [(n, c.dtype) for n, c in
                                 zip(chunk.names, chunk.columns)])
    with ipc.new_file(target, tbl.schema) as w:
        w.write_table(tbl)
    return target
