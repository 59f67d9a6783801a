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
        if "://" in p and not p.startswith("file://"):
            from ..storage.object_store import expand_to_local

            out.extend(expand_to_local(p))
            continue
        if p.startswith("file://"):
            p = p[len("file://"):]
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


def _partition_info(root: str, files: List[str]):
    """Hive-style `key=value` path partitioning (ref: sail-data-source
    src/listing/ partition handling): per-file partition values, inferred
    int64 where every value parses, else string."""
    import urllib.parse

    keys: List[str] = []
    per_file: List[dict] = []
    for f in files:
        rel = os.path.relpath(f, root)
        vals = {}
        for seg in rel.split(os.sep)[:-1]:
            if "=" in seg:
                k, _, v = seg.partition("=")
                vals[k] = urllib.parse.unquote(v)
                if k not in keys:
                    keys.append(k)
        per_file.append(vals)
    if not keys or any(len(v) != len(keys) for v in per_file):
        return [], []
    types = []
    for k in keys:
        try:
            for v in per_file:
                int(v[k])
            types.append(T.I64)
        except ValueError:
            types.append(T.STRING)
    return list(zip(keys, types)), per_file


def infer_schema(paths: List[str]) -> List[Tuple[str, T.DataType]]:
    files = _expand(paths)
    if not files:
        raise FileNotFoundError(f"no parquet files under {paths}")
    sch = pq.read_schema(files[0])
    out = [(f.name, _engine_type(f.type)) for f in sch]
    if len(paths) == 1 and os.path.isdir(paths[0]):
        pkeys, _ = _partition_info(paths[0], files)
        out += pkeys
    return out


def _gpu_mode(options: Dict[str, str]) -> str:
    if options and options.get("gpuDecode") is not None:
        return str(options["gpuDecode"]).lower()
    return os.environ.get("SAIL_IO_GPU_PARQUET", "auto").lower()


def _try_read_gpu(files: List[str], schema, device, options):
    """GPU page-decode path (gpu_parquet.py); None => host fallback."""
    mode = _gpu_mode(options)
    if mode in ("off", "false", "0") or not str(device).startswith("cuda"):
        if mode == "force":
            raise RuntimeError("gpuDecode=force but device is not cuda")
        return None
    from . import gpu_parquet

    try:
        return gpu_parquet.read_gpu(files, schema, device)
    except gpu_parquet.Unsupported:
        if mode == "force":
            raise
        return None


def read(paths: List[str], schema, device, options: Dict[str, str]):
    files = _expand(paths)
    pkeys, pvals = ([], [])
    if len(paths) == 1 and os.path.isdir(paths[0]):
        pkeys, pvals = _partition_info(paths[0], files)
    if not pkeys:
        t = _try_read_gpu(files, schema, device, options)
        if t is not None:
            return t
        cols = [n for n, _ in schema] if schema else None
        tbl = pq.read_table(files, columns=cols)
        return arrow_to_table(tbl, device=device)
    # partition pruning: equality filters pushed down as options
    # "partition.<key>" (see plan/rules/pushdown.py)
    if options:
        for k, t in pkeys:
            want = options.get(f"partition.{k}")
            if want is not None:
                def _match(raw, want=want, t=t):
                    if t == T.I64:
                        try:
                            return int(raw) == int(float(want))
                        except (ValueError, TypeError):
                            return False
                    return raw == str(want)

                keep = [i for i, v in enumerate(pvals) if _match(v[k])]
                files = [files[i] for i in keep]
                pvals = [pvals[i] for i in keep]
    data_names = {f.name for f in pq.read_schema(files[0])} if files else set()
    from ..engine.column import Column, StringColumn, Table

    parts = []
    dschema = [(n, t) for n, t in schema if n in data_names] if schema else None
    for f, v in zip(files, pvals):
        gt = _try_read_gpu([f], dschema, device, options)
        if gt is not None:
            parts.append((gt, v))
            continue
        t = pq.read_table(f, columns=[n for n, _ in dschema] if dschema else None)
        parts.append((arrow_to_table(t, device=device), v))
    if not parts:
        cols = {}
        for n, t in (schema or []):
            cols[n] = Column.from_values([], t, device=device)
        return Table(cols)
    from ..engine.executor import concat_columns

    tables = [t for t, _ in parts]  # already engine Tables (GPU or host path)
    out_cols = {}
    for n in tables[0].columns:
        out_cols[n] = concat_columns([t.columns[n] for t in tables])             if len(tables) > 1 else tables[0].columns[n]
    for k, kt in pkeys:
        vals = []
        for (t, v) in parts:
            pv = int(v[k]) if kt == T.I64 else v[k]
            vals.extend([pv] * t.num_rows)
        out_cols[k] = Column.from_values(vals, kt, device=device)
    return Table(out_cols)


def write(path: str, chunk, mode: str, options: Dict[str, str]):
    schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
    tbl = chunk_to_arrow(chunk, schema)
    if "://" in path and not path.startswith("file://"):
        # object-store write: encode to a buffer, put via the registry
        from ..storage.object_store import global_registry

        store, sp = global_registry().for_uri(path)
        if not sp.endswith(".parquet"):
            sp = sp.rstrip("/") + "/part-00000.parquet"
        if mode == "error" and store.exists(sp):
            raise FileExistsError(path)
        import io as _io

        sink = _io.BytesIO()
        pq.write_table(tbl, sink,
                       compression=options.get("compression", "snappy"))
        store.write_bytes(sp, sink.getvalue())
        return path
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


def scan_batches(paths: List[str], schema, device, options: Dict[str, str],
                 target_rows: int = 8_000_000):
    """Out-of-core scan: yield engine Tables in bounded row batches (one or
    more row groups at a time) instead of materializing whole files — the
    spill-free analogue of the reference's bounded-memory streams
    (ref: application.yaml runtime memory pools / execution.batch_size).
    Uses the GPU page decoder per batch when supported."""
    files = _expand(paths)
    mode = _gpu_mode(options)
    use_gpu = mode not in ("off", "false", "0") and str(device).startswith("cuda")
    from . import gpu_parquet

    for f in files:
        pf = pq.ParquetFile(f)
        nrg = pf.metadata.num_row_groups
        rows_per_rg = max(pf.metadata.num_rows // max(nrg, 1), 1)
        step = max(1, target_rows // rows_per_rg)
        for lo in range(0, nrg, step):
            hi = min(lo + step, nrg)
            if use_gpu:
                try:
                    yield gpu_parquet.read_gpu([f], schema, device,
                                               rg_window=(lo, hi))
                    continue
                except gpu_parquet.Unsupported:
                    if mode == "force":
                        raise
            cols = [n for n, _ in schema] if schema else None
            tbl = pf.read_row_groups(list(range(lo, hi)), columns=cols)
            yield arrow_to_table(tbl, device=device)
