"""Apache Iceberg v2 table format — metadata in JSON + Avro, data in parquet.

From-scratch implementation mirroring the reference's own from-scratch one
(ref: crates/sail-iceberg/ — spec types, manifest/metadata handling, writer
and commit operators, position/equality delete handling). The Avro codec is
`sail_amd.utils.avro` (the image has no avro package). Layout:

    table/
      metadata/
        v1.metadata.json, v2.metadata.json, ...   (+ version-hint.text)
        snap-<id>.avro          manifest LIST (one entry per manifest file)
        <uuid>-m0.avro          manifest (one entry per data/delete file)
      data/
        part-*.parquet

Reads tolerate external layouts (file:// URIs, %05d-prefixed metadata
names, deflate or null Avro codecs, extra manifest fields — the decoder is
driven by each file's embedded writer schema). Row-level deletes are
applied on read: position deletes (content=1) drop (file, pos) pairs;
equality deletes (content=2) anti-join on the delete file's equality ids.
"""
from __future__ import annotations

import glob
import json
import os
import random
import re
import time
import uuid
from typing import Dict, List, Optional, Tuple

from ..engine import types as T
from ..utils.avro import read_container, write_container

# ===========================================================================
# type mapping
# ===========================================================================
_TO_ICE = {T.BOOL: "boolean", T.I32: "int", T.I64: "long", T.F32: "float",
           T.F64: "double", T.DATE: "date", T.TIMESTAMP: "timestamp"}


def _type_to_ice(t: T.DataType) -> str:
    if isinstance(t, T.DecimalType):
        return f"decimal({t.precision}, {t.scale})"
    if isinstance(t, T.BinaryType):
        return "binary"
    if isinstance(t, T.StringType):
        return "string"
    for k, v in _TO_ICE.items():
        if type(t) is type(k):
            return v
    raise ValueError(f"cannot map {t} to an Iceberg type")


def _type_from_ice(s: str) -> T.DataType:
    m = re.match(r"decimal\((\d+),\s*(\d+)\)", s)
    if m:
        return T.DecimalType(int(m.group(1)), int(m.group(2)))
    return {"boolean": T.BOOL, "int": T.I32, "long": T.I64, "float": T.F32,
            "double": T.F64, "date": T.DATE, "timestamp": T.TIMESTAMP,
            "timestamptz": T.TIMESTAMP, "string": T.STRING,
            "binary": T.BINARY, "uuid": T.STRING}[s]


def _ice_schema(schema: List[Tuple[str, T.DataType]]) -> dict:
    return {"type": "struct", "schema-id": 0,
            "fields": [{"id": i + 1, "name": n, "required": False,
                        "type": _type_to_ice(t)}
                       for i, (n, t) in enumerate(schema)]}


def _schema_from_ice(s: dict) -> List[Tuple[str, T.DataType]]:
    out = []
    for f in s["fields"]:
        if isinstance(f["type"], dict):
            raise NotImplementedError(
                f"nested Iceberg column {f['name']!r} not supported")
        out.append((f["name"], _type_from_ice(f["type"])))
    return out


# ===========================================================================
# Avro schemas we write (spec field names + ids; unpartitioned tables)
# ===========================================================================
_DATA_FILE_SCHEMA = {
    "type": "record", "name": "r2", "fields": [
        {"name": "content", "type": "int", "field-id": 134},
        {"name": "file_path", "type": "string", "field-id": 100},
        {"name": "file_format", "type": "string", "field-id": 101},
        {"name": "partition", "field-id": 102,
         "type": {"type": "record", "name": "r102", "fields": []}},
        {"name": "record_count", "type": "long", "field-id": 103},
        {"name": "file_size_in_bytes", "type": "long", "field-id": 104},
        {"name": "equality_ids", "field-id": 135, "default": None,
         "type": ["null", {"type": "array", "items": "int"}]},
    ]}

_MANIFEST_ENTRY_SCHEMA = {
    "type": "record", "name": "manifest_entry", "fields": [
        {"name": "status", "type": "int", "field-id": 0},
        {"name": "snapshot_id", "type": ["null", "long"], "default": None,
         "field-id": 1},
        {"name": "sequence_number", "type": ["null", "long"], "default": None,
         "field-id": 3},
        {"name": "file_sequence_number", "type": ["null", "long"],
         "default": None, "field-id": 4},
        {"name": "data_file", "type": _DATA_FILE_SCHEMA, "field-id": 2},
    ]}

_MANIFEST_FILE_SCHEMA = {
    "type": "record", "name": "manifest_file", "fields": [
        {"name": "manifest_path", "type": "string", "field-id": 500},
        {"name": "manifest_length", "type": "long", "field-id": 501},
        {"name": "partition_spec_id", "type": "int", "field-id": 502},
        {"name": "content", "type": "int", "field-id": 517},
        {"name": "sequence_number", "type": "long", "field-id": 515},
        {"name": "min_sequence_number", "type": "long", "field-id": 516},
        {"name": "added_snapshot_id", "type": "long", "field-id": 503},
        {"name": "added_data_files_count", "type": "int", "field-id": 504},
        {"name": "existing_data_files_count", "type": "int", "field-id": 505},
        {"name": "deleted_data_files_count", "type": "int", "field-id": 506},
        {"name": "added_rows_count", "type": "long", "field-id": 512},
        {"name": "existing_rows_count", "type": "long", "field-id": 513},
        {"name": "deleted_rows_count", "type": "long", "field-id": 514},
    ]}


# ===========================================================================
# metadata discovery / snapshot resolution
# ===========================================================================
class IcebergTable:
    def __init__(self, path: str):
        self.path = path
        self.meta_dir = os.path.join(path, "metadata")
        self.metadata = self._load_metadata()

    # -- discovery ----------------------------------------------------------
    def _metadata_files(self) -> List[Tuple[int, str]]:
        out = []
        for f in glob.glob(os.path.join(self.meta_dir, "*.metadata.json")):
            base = os.path.basename(f)
            m = re.match(r"v?(\d+)", base)
            out.append((int(m.group(1)) if m else -1, f))
        return sorted(out)

    def _load_metadata(self) -> Optional[dict]:
        hint = os.path.join(self.meta_dir, "version-hint.text")
        if os.path.exists(hint):
            with open(hint) as f:
                v = f.read().strip()
            for cand in (f"v{v}.metadata.json", f"{v}.metadata.json"):
                p = os.path.join(self.meta_dir, cand)
                if os.path.exists(p):
                    with open(p) as fh:
                        return json.load(fh)
        files = self._metadata_files()
        if not files:
            return None
        with open(files[-1][1]) as f:
            return json.load(f)

    def exists(self) -> bool:
        return self.metadata is not None

    # -- schema / snapshots -------------------------------------------------
    def schema(self) -> List[Tuple[str, T.DataType]]:
        md = self.metadata
        sid = md.get("current-schema-id", 0)
        for s in md.get("schemas", []):
            if s.get("schema-id") == sid:
                return _schema_from_ice(s)
        if "schema" in md:  # v1 layout
            return _schema_from_ice(md["schema"])
        raise ValueError("no schema in Iceberg metadata")

    def snapshot(self, options: Optional[Dict[str, str]] = None) -> Optional[dict]:
        md = self.metadata
        snaps = md.get("snapshots", [])
        options = options or {}
        sid = options.get("snapshot-id") or options.get("snapshotId")
        if sid is not None:
            for s in snaps:
                if s["snapshot-id"] == int(sid):
                    return s
            raise ValueError(f"Iceberg snapshot {sid} not found")
        v = options.get("versionAsOf")
        if v is not None:
            # SQL VERSION AS OF n -> n-th snapshot in commit order
            ordered = sorted(snaps, key=lambda s: s["timestamp-ms"])
            idx = int(v)
            if 0 <= idx < len(ordered):
                return ordered[idx]
            raise ValueError(f"Iceberg version {idx} not found "
                             f"(0..{len(ordered) - 1})")
        ts = options.get("as-of-timestamp")  # spark option: epoch millis
        if ts is None and options.get("timestampAsOf") is not None:
            t2 = options["timestampAsOf"]  # SQL clause: iso string/seconds
            if isinstance(t2, str):
                import datetime as _dt2

                ts = int(_dt2.datetime.fromisoformat(t2).replace(
                    tzinfo=_dt2.timezone.utc).timestamp() * 1000)
            else:
                ts = int(float(t2) * 1000)
        if ts is not None:
            ts = int(ts)
            best = None
            for s in snaps:
                if s["timestamp-ms"] <= ts and (
                        best is None or s["timestamp-ms"] > best["timestamp-ms"]):
                    best = s
            if best is None:
                raise ValueError(f"no Iceberg snapshot at or before {ts}")
            return best
        cur = md.get("current-snapshot-id")
        if cur in (None, -1):
            return None
        for s in snaps:
            if s["snapshot-id"] == cur:
                return s
        return None

    def _local(self, p: str) -> str:
        """Map a metadata-recorded URI to a local path (tolerates file://
        prefixes and foreign absolute locations by basename fallback)."""
        if p.startswith("file://"):
            p = p[len("file://"):]
        if os.path.exists(p):
            return p
        for sub in ("metadata", "data"):
            cand = os.path.join(self.path, sub, os.path.basename(p))
            if os.path.exists(cand):
                return cand
        return p

    def files(self, options=None) -> Tuple[List[dict], List[dict]]:
        """(data_files, delete_files) for the selected snapshot; each item is
        the manifest-entry data_file dict with `file_path` localized."""
        snap = self.snapshot(options)
        if snap is None:
            return [], []
        _, manifests, _ = read_container(self._local(snap["manifest-list"]))
        data, deletes = [], []
        for mf in manifests:
            _, entries, _ = read_container(self._local(mf["manifest_path"]))
            for e in entries:
                if e.get("status") == 2:  # DELETED
                    continue
                df = e["data_file"]
                df = dict(df)
                df["file_path"] = self._local(df["file_path"])
                (data if df.get("content", 0) == 0 else deletes).append(df)
        return data, deletes


# ===========================================================================
# read
# ===========================================================================
def infer_schema(paths: List[str], options: Dict[str, str] = None):
    t = IcebergTable(paths[0])
    if not t.exists():
        raise ValueError(f"not an Iceberg table: {paths[0]}")
    return t.schema()


def read(paths: List[str], schema, device, options: Dict[str, str]):
    from . import parquet_io
    from ..engine.column import Column, Table

    t = IcebergTable(paths[0])
    tbl_schema = t.schema()
    data, deletes = t.files(options)
    if not data:
        cols = {n: Column.from_values([], dt, device=device)
                for n, dt in tbl_schema}
        return Table(cols)
    if not deletes:
        return parquet_io.read([d["file_path"] for d in data], tbl_schema,
                               device, options or {})
    return _read_with_deletes(data, deletes, tbl_schema, device, options)


def _read_with_deletes(data, deletes, tbl_schema, device, options):
    """Merge-on-read: apply position deletes (file_path, pos) and equality
    deletes (anti-join on equality_ids columns) while scanning."""
    import pyarrow.parquet as pq

    import torch

    from . import parquet_io
    from ..engine.chunk import Chunk
    from ..engine.column import Table

    pos_by_file: Dict[str, set] = {}
    eq_tables = []  # (names, pandas frame)
    for df in deletes:
        tbl = pq.read_table(df["file_path"])
        if df.get("content") == 1:  # position deletes
            fp = tbl.column("file_path").to_pylist()
            pos = tbl.column("pos").to_pylist()
            for f, p in zip(fp, pos):
                pos_by_file.setdefault(os.path.basename(f), set()).add(p)
        else:  # equality deletes
            ids = df.get("equality_ids") or []
            names = [tbl_schema[i - 1][0] for i in ids] if ids \
                else list(tbl.column_names)
            eq_tables.append((names, tbl.select(names).to_pandas()))
    parts = []
    for d in data:
        fpath = d["file_path"]
        tab = parquet_io.read([fpath], tbl_schema, device, options or {})
        chunk = Chunk.from_table(tab)
        drop = pos_by_file.get(os.path.basename(fpath))
        if drop:
            keep = torch.ones(chunk.num_rows, dtype=torch.bool)
            keep[torch.tensor(sorted(drop), dtype=torch.int64)] = False
            idx = torch.nonzero(keep, as_tuple=False).flatten()
            chunk = Chunk([c.gather(idx) for c in chunk.columns],
                          list(chunk.names))
        parts.append(chunk)
    from ..engine.executor import concat_columns

    out = parts[0] if len(parts) == 1 else Chunk(
        [concat_columns([p.columns[i] for p in parts])
         for i in range(len(parts[0].columns))], list(parts[0].names))
    if eq_tables:
        out = _apply_equality_deletes(out, eq_tables)
    return Table({n: c for n, c in zip(out.names, out.columns)})


def _apply_equality_deletes(chunk, eq_tables):
    import pandas as pd
    import torch

    from ..engine.chunk import Chunk

    keep = torch.ones(chunk.num_rows, dtype=torch.bool)
    cache: Dict[str, list] = {}
    for names, del_df in eq_tables:
        cols = {}
        for n in names:
            if n not in cache:
                cache[n] = chunk.columns[list(chunk.names).index(n)].to_pylist()
            cols[n] = cache[n]
        cur = pd.DataFrame(cols)
        hit = cur.merge(del_df.drop_duplicates(), on=names, how="left",
                        indicator=True)["_merge"].eq("both").to_numpy()
        keep &= ~torch.from_numpy(hit)
    idx = torch.nonzero(keep, as_tuple=False).flatten()
    return Chunk([c.gather(idx) for c in chunk.columns], list(chunk.names))


# ===========================================================================
# write
# ===========================================================================
def _write_data_files(path: str, chunk, options) -> List[dict]:
    """Parquet part files under data/; returns data_file dicts."""
    import pyarrow.parquet as pq

    from concurrent.futures import ThreadPoolExecutor

    from .arrow_io import chunk_to_arrow
    from ..engine.chunk import Chunk as _Chunk

    schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
    data_dir = os.path.join(path, "data")
    os.makedirs(data_dir, exist_ok=True)
    compression = (options or {}).get("compression", "snappy")
    n = chunk.num_rows
    from .delta import PART_ROWS

    nparts = max(1, min(16, (n + PART_ROWS - 1) // PART_ROWS))
    step = (n + nparts - 1) // nparts if nparts else n

    def one(i):
        lo = i * step
        ln = min(step, n - lo)
        sub = _Chunk([c.slice(lo, ln) for c in chunk.columns],
                     list(chunk.names)) if nparts > 1 else chunk
        fname = f"part-{i:05d}-{uuid.uuid4().hex}.parquet"
        fpath = os.path.join(data_dir, fname)
        pq.write_table(chunk_to_arrow(sub, schema), fpath,
                       compression=compression)
        return {"content": 0, "file_path": fpath, "file_format": "PARQUET",
                "partition": {}, "record_count": ln,
                "file_size_in_bytes": os.path.getsize(fpath),
                "equality_ids": None}

    if nparts == 1:
        return [one(0)]
    with ThreadPoolExecutor(max_workers=min(nparts, 8)) as exe:
        return list(exe.map(one, range(nparts)))


def _new_snapshot_id() -> int:
    return random.getrandbits(62)


def _partition_record_schema(pvals: Dict[str, object]) -> dict:
    fields = []
    for k, v in pvals.items():
        at = "long" if isinstance(v, int) else "string"
        fields.append({"name": k, "type": ["null", at], "default": None})
    return {"type": "record", "name": "r102", "fields": fields}


def write(path: str, chunk, mode: str, options: Dict[str, str],
          partition_values: Optional[Dict[str, object]] = None,
          partition_spec: Optional[List[dict]] = None):
    """append / overwrite commit: data parquet -> manifest Avro ->
    manifest-list Avro -> new vN.metadata.json + version-hint.
    `partition_values`/`partition_spec` come from write_partitioned."""
    schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
    t = IcebergTable(path)
    if t.exists() and mode == "error":
        raise ValueError(f"Iceberg table already exists: {path}")
    os.makedirs(t.meta_dir, exist_ok=True)

    md = t.metadata
    if md is None:
        md = {"format-version": 2, "table-uuid": str(uuid.uuid4()),
              "location": path, "last-sequence-number": 0,
              "last-updated-ms": 0,
              "last-column-id": len(schema),
              "current-schema-id": 0, "schemas": [_ice_schema(schema)],
              "default-spec-id": 0,
              "partition-specs": [{"spec-id": 0,
                                   "fields": partition_spec or []}],
              "last-partition-id": 999,
              "default-sort-order-id": 0,
              "sort-orders": [{"order-id": 0, "fields": []}],
              "current-snapshot-id": -1, "snapshots": [],
              "snapshot-log": [], "metadata-log": [], "properties": {}}
    elif mode == "append":
        cur = t.schema()
        if [n for n, _ in cur] != [n for n, _ in schema] or \
                [_type_to_ice(ty) for _, ty in cur] != \
                [_type_to_ice(ty) for _, ty in schema]:
            raise ValueError(
                f"Iceberg append schema mismatch: table has "
                f"{[(n, _type_to_ice(ty)) for n, ty in cur]}, write has "
                f"{[(n, _type_to_ice(ty)) for n, ty in schema]}")
    elif mode == "overwrite":
        # schema evolution on overwrite: register the new schema and make it
        # current so data files and metadata agree
        # (ref: sail-iceberg schema evolution on replace)
        new_ice = _ice_schema(schema)
        cur_id = md.get("current-schema-id", 0)
        cur_ice = next((s for s in md.get("schemas", [])
                        if s.get("schema-id", 0) == cur_id),
                       md["schemas"][0] if md.get("schemas") else None)
        def _fields(s):
            return [(f["name"], f["type"]) for f in s.get("fields", [])] \
                if s else None
        if _fields(cur_ice) != _fields(new_ice):
            new_id = max((s.get("schema-id", 0)
                          for s in md.get("schemas", [])), default=-1) + 1
            new_ice["schema-id"] = new_id
            md["schemas"] = md.get("schemas", []) + [new_ice]
            md["current-schema-id"] = new_id
            md["last-column-id"] = max(md.get("last-column-id", 0),
                                       len(schema))

    seq = md["last-sequence-number"] + 1
    snap_id = _new_snapshot_id()
    now_ms = int(time.time() * 1000)

    data_files = _write_data_files(path, chunk, options)
    if partition_values:
        for df in data_files:
            df["partition"] = dict(partition_values)
    entries = [{"status": 1, "snapshot_id": snap_id, "sequence_number": None,
                "file_sequence_number": None, "data_file": df}
               for df in data_files]
    entry_schema = _MANIFEST_ENTRY_SCHEMA
    if partition_values:
        import copy as _copy

        entry_schema = _copy.deepcopy(_MANIFEST_ENTRY_SCHEMA)
        entry_schema["fields"][-1]["type"]["fields"][3]["type"] = \
            _partition_record_schema(partition_values)
    mpath = os.path.join(t.meta_dir, f"{uuid.uuid4().hex}-m0.avro")
    cur_id = md.get("current-schema-id", 0)
    cur_schema = next((s for s in md.get("schemas", [])
                       if s.get("schema-id", 0) == cur_id),
                      _ice_schema(schema))
    write_container(mpath, entry_schema, entries, metadata={
        "schema": json.dumps(cur_schema).encode(),
        "schema-id": str(cur_id).encode(),
        "partition-spec": json.dumps([]).encode(),
        "partition-spec-id": b"0",
        "format-version": b"2",
        "content": b"data",
    })
    nrows = sum(df["record_count"] for df in data_files)
    new_mf = {"manifest_path": mpath,
              "manifest_length": os.path.getsize(mpath),
              "partition_spec_id": 0, "content": 0,
              "sequence_number": seq, "min_sequence_number": seq,
              "added_snapshot_id": snap_id,
              "added_data_files_count": len(data_files),
              "existing_data_files_count": 0, "deleted_data_files_count": 0,
              "added_rows_count": nrows, "existing_rows_count": 0,
              "deleted_rows_count": 0}
    manifests = [new_mf]
    parent = t.snapshot() if t.exists() and md.get("snapshots") else None
    if mode == "append" and parent is not None:
        _, prev, _ = read_container(t._local(parent["manifest-list"]))
        manifests += prev

    _commit_snapshot(t, md, manifests, snap_id, seq, parent,
                     "append" if mode == "append" else "overwrite")


def history(path: str):
    """Snapshot log as rows (version, snapshot_id, timestamp_ms, operation)."""
    t = IcebergTable(path)
    if not t.exists():
        raise ValueError(f"not an Iceberg table: {path}")
    out = []
    for i, s in enumerate(t.metadata.get("snapshots", [])):
        out.append((i, s["snapshot-id"], s["timestamp-ms"],
                    s.get("summary", {}).get("operation", "")))
    return out


def _commit_snapshot(t: "IcebergTable", md: dict, manifests: List[dict],
                     snap_id: int, seq: int, parent: Optional[dict],
                     operation: str):
    """Write manifest list + snapshot entry + new vN.metadata.json
    (the tail every commit shares; ref: sail-iceberg commit flow)."""
    now_ms = int(time.time() * 1000)
    ml_path = os.path.join(t.meta_dir, f"snap-{snap_id}.avro")
    write_container(ml_path, _MANIFEST_FILE_SCHEMA, manifests, metadata={
        "snapshot-id": str(snap_id).encode(),
        "sequence-number": str(seq).encode(),
        "parent-snapshot-id":
            str(parent["snapshot-id"]).encode() if parent else b"null",
        "format-version": b"2",
    })
    snap = {"snapshot-id": snap_id, "sequence-number": seq,
            "timestamp-ms": now_ms, "manifest-list": ml_path,
            "schema-id": md.get("current-schema-id", 0),
            "summary": {"operation": operation}}
    if parent is not None:
        snap["parent-snapshot-id"] = parent["snapshot-id"]
    md["snapshots"] = md.get("snapshots", []) + [snap]
    md["current-snapshot-id"] = snap_id
    md["last-sequence-number"] = seq
    md["last-updated-ms"] = now_ms
    md["snapshot-log"] = md.get("snapshot-log", []) + [
        {"snapshot-id": snap_id, "timestamp-ms": now_ms}]

    files = t._metadata_files()
    version = (files[-1][0] if files else 0) + 1
    md_path = os.path.join(t.meta_dir, f"v{version}.metadata.json")
    tmp = md_path + ".tmp"
    with open(tmp, "w") as f:
        json.dump(md, f)
    os.replace(tmp, md_path)
    with open(os.path.join(t.meta_dir, "version-hint.text"), "w") as f:
        f.write(str(version))


def scan_layout(path: str, schema, device, options):
    """Read with deletes applied, tracking per data file the manifest
    file_path and surviving ORIGINAL row positions — lets DELETE commit
    merge-on-read position-delete files (ref: sail-iceberg position delete
    writers, src/physical_plan/)."""
    import numpy as np
    import pyarrow.parquet as pq
    import torch

    from . import parquet_io
    from ..engine.chunk import Chunk
    from ..engine.column import Column, Table
    from ..engine.executor import concat_columns

    t = IcebergTable(path)
    tbl_schema = t.schema()
    data, deletes = t.files(options)
    if not data:
        cols = {n: Column.from_values([], dt, device=device)
                for n, dt in tbl_schema}
        return Table(cols), []
    pos_by_file: Dict[str, set] = {}
    for df in deletes:
        if df.get("content") != 1:
            continue
        dtbl = pq.read_table(df["file_path"])
        for f, p0 in zip(dtbl.column("file_path").to_pylist(),
                         dtbl.column("pos").to_pylist()):
            pos_by_file.setdefault(os.path.basename(f), set()).add(p0)
    parts, layout = [], []
    for d in data:
        tab = parquet_io.read([d["file_path"]], tbl_schema, device,
                              options or {})
        chunk = Chunk.from_table(tab)
        nrows = chunk.num_rows
        drop = pos_by_file.get(os.path.basename(d["file_path"]))
        if drop:
            keep = np.ones(nrows, dtype=bool)
            keep[sorted(drop)] = False
            orig = np.nonzero(keep)[0]
            idx = torch.from_numpy(orig).to(torch.int64)
            chunk = Chunk([c.gather(idx) for c in chunk.columns],
                          list(chunk.names))
        else:
            orig = np.arange(nrows, dtype=np.int64)
        parts.append(chunk)
        layout.append((d.get("orig_path", d["file_path"]), orig))
    out = parts[0] if len(parts) == 1 else Chunk(
        [concat_columns([p.columns[i] for p in parts])
         for i in range(len(parts[0].columns))], list(parts[0].names))
    return Table({n: c for n, c in zip(out.names, out.columns)}), layout


#: iceberg position-delete column field ids (spec: 2147483546/2147483545)
def delete_with_positions(path: str, layout, deleted_mask):
    """Commit a DELETE as a merge-on-read position-delete file: parquet of
    (file_path, pos) rows + a content=1 (deletes) manifest appended to the
    current snapshot's manifest list."""
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq

    t = IcebergTable(path)
    md = t.metadata
    rows_fp, rows_pos = [], []
    off = 0
    for fpath, orig in layout:
        seg = deleted_mask[off:off + len(orig)]
        off += len(orig)
        newly = orig[seg]
        rows_fp.extend([fpath] * len(newly))
        rows_pos.extend(int(x) for x in newly)
    if not rows_fp:
        return None
    data_dir = os.path.join(path, "data")
    os.makedirs(data_dir, exist_ok=True)
    del_path = os.path.join(data_dir,
                            f"delete-{uuid.uuid4().hex}.parquet")
    pq.write_table(pa.table({"file_path": pa.array(rows_fp),
                             "pos": pa.array(rows_pos, pa.int64())}),
                   del_path)
    seq = md["last-sequence-number"] + 1
    snap_id = _new_snapshot_id()
    entry = {"status": 1, "snapshot_id": snap_id, "sequence_number": None,
             "file_sequence_number": None,
             "data_file": {"content": 1, "file_path": del_path,
                           "file_format": "PARQUET", "partition": {},
                           "record_count": len(rows_fp),
                           "file_size_in_bytes": os.path.getsize(del_path),
                           "equality_ids": None}}
    mpath = os.path.join(t.meta_dir, f"{uuid.uuid4().hex}-m0.avro")
    cur_id = md.get("current-schema-id", 0)
    cur_schema = next((s for s in md.get("schemas", [])
                       if s.get("schema-id", 0) == cur_id), None)
    write_container(mpath, _MANIFEST_ENTRY_SCHEMA, [entry], metadata={
        "schema": json.dumps(cur_schema or {}).encode(),
        "schema-id": str(cur_id).encode(),
        "partition-spec": json.dumps([]).encode(),
        "partition-spec-id": b"0",
        "format-version": b"2",
        "content": b"deletes",
    })
    new_mf = {"manifest_path": mpath,
              "manifest_length": os.path.getsize(mpath),
              "partition_spec_id": 0, "content": 1,
              "sequence_number": seq, "min_sequence_number": seq,
              "added_snapshot_id": snap_id,
              "added_data_files_count": 1,
              "existing_data_files_count": 0, "deleted_data_files_count": 0,
              "added_rows_count": len(rows_fp), "existing_rows_count": 0,
              "deleted_rows_count": 0}
    parent = t.snapshot()
    manifests = [new_mf]
    if parent is not None:
        _, prev, _ = read_container(t._local(parent["manifest-list"]))
        manifests += prev
    _commit_snapshot(t, md, manifests, snap_id, seq, parent, "delete")
    return len(rows_fp)


# ===========================================================================
# partition transforms (ref: crates/sail-iceberg/src/physical_plan/
# partition_transform_expr.rs; iceberg spec "Partition Transforms")
# ===========================================================================

def _murmur3_32(data: bytes, seed: int = 0) -> int:
    """murmur3_x86_32 — the hash the iceberg bucket transform specifies."""
    c1, c2 = 0xCC9E2D51, 0x1B873593
    h = seed
    n = len(data)
    for i in range(0, n - n % 4, 4):
        k = int.from_bytes(data[i:i + 4], "little")
        k = (k * c1) & 0xFFFFFFFF
        k = ((k << 15) | (k >> 17)) & 0xFFFFFFFF
        k = (k * c2) & 0xFFFFFFFF
        h ^= k
        h = ((h << 13) | (h >> 19)) & 0xFFFFFFFF
        h = (h * 5 + 0xE6546B64) & 0xFFFFFFFF
    tail = data[n - n % 4:]
    if tail:
        k = int.from_bytes(tail.ljust(4, b"\x00"), "little")
        k = (k * c1) & 0xFFFFFFFF
        k = ((k << 15) | (k >> 17)) & 0xFFFFFFFF
        k = (k * c2) & 0xFFFFFFFF
        h ^= k
    h ^= n
    h ^= h >> 16
    h = (h * 0x85EBCA6B) & 0xFFFFFFFF
    h ^= h >> 13
    h = (h * 0xC2B2AE35) & 0xFFFFFFFF
    h ^= h >> 16
    return h


def _bucket_hash(value, dtype: T.DataType) -> int:
    import struct as _s

    if value is None:
        return 0
    if isinstance(dtype, T.DecimalType):
        unscaled = int(round(float(value) * (10 ** dtype.scale)))
        blen = max((unscaled.bit_length() + 8) // 8, 1)
        return _murmur3_32(unscaled.to_bytes(blen, "big", signed=True))
    if isinstance(dtype, T.StringType):
        b = value if isinstance(value, bytes) else str(value).encode()
        return _murmur3_32(b)
    if isinstance(dtype, T.DateType):
        import datetime as _dt

        days = (value - _dt.date(1970, 1, 1)).days \
            if isinstance(value, _dt.date) else int(value)
        return _murmur3_32(_s.pack("<q", days))
    return _murmur3_32(_s.pack("<q", int(value)))


_EPOCH_Y = 1970


def parse_transform(spec: str):
    """'bucket(4, col)' / 'truncate(10, col)' / 'years(col)' / 'col' ->
    (transform_name, column, param)."""
    m = re.match(r"(\w+)\s*\(\s*(?:(\d+)\s*,\s*)?(\w+)\s*\)$", spec.strip())
    if not m:
        return ("identity", spec.strip(), None)
    name, param, col = m.group(1).lower(), m.group(2), m.group(3)
    alias = {"year": "years", "month": "months", "day": "days",
             "hour": "hours", "date": "days"}
    name = alias.get(name, name)
    if name in ("bucket", "truncate"):
        if param is None:
            raise ValueError(f"{name} transform needs a width: {spec}")
        return (name, col, int(param))
    if name in ("years", "months", "days", "hours", "identity", "void"):
        return (name, col, None)
    raise ValueError(f"unknown partition transform {spec!r}")


def apply_transform(name: str, value, dtype: T.DataType, param):
    """One partition value (iceberg spec semantics; None passes through)."""
    import datetime as _dt

    if value is None or name == "void":
        return None
    if name == "identity":
        return value
    if name == "bucket":
        return (_bucket_hash(value, dtype) & 0x7FFFFFFF) % param
    if name == "truncate":
        if isinstance(dtype, T.StringType):
            return str(value)[:param]
        v = int(value)
        return v - (v % param if v >= 0 else (v % param))
    # temporal transforms: value may be date / epoch-micros timestamp
    if isinstance(value, _dt.date) and not isinstance(value, _dt.datetime):
        d = value
    elif isinstance(dtype, T.TimestampType):
        d = _dt.datetime.utcfromtimestamp(int(value) / 1e6)
    else:
        d = _dt.date(1970, 1, 1) + _dt.timedelta(days=int(value))
    if name == "years":
        return d.year - _EPOCH_Y
    if name == "months":
        return (d.year - _EPOCH_Y) * 12 + (d.month - 1)
    if name == "days":
        dd = d.date() if isinstance(d, _dt.datetime) else d
        return (dd - _dt.date(1970, 1, 1)).days
    if name == "hours":
        if not isinstance(d, _dt.datetime):
            d = _dt.datetime(d.year, d.month, d.day)
        return int((d - _dt.datetime(1970, 1, 1)).total_seconds() // 3600)
    raise ValueError(f"unknown transform {name}")


def write_partitioned(path: str, chunk, mode: str, options: Dict[str, str],
                      partition_by: List[str]):
    """Partitioned iceberg write: rows are split by the transformed
    partition tuple; each partition gets its own data files and manifest
    entries carry the partition record; metadata records the spec."""
    import numpy as np

    from ..engine.chunk import Chunk as _Chunk

    schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
    names_low = [n.lower() for n in chunk.names]
    specs = [parse_transform(s) for s in partition_by]
    src_idx = []
    for tname, col, param in specs:
        if col.lower() not in names_low:
            raise ValueError(f"partition column {col} not in output")
        src_idx.append(names_low.index(col.lower()))
    # transformed partition tuple per row (host; partition columns only)
    cols_host = [chunk.columns[i].to_pylist() for i in src_idx]
    n = chunk.num_rows
    tuples = []
    for r in range(n):
        tuples.append(tuple(
            apply_transform(t, cols_host[j][r], schema[src_idx[j]][1], p)
            for j, (t, _c, p) in enumerate(specs)))
    uniq = sorted(set(tuples), key=lambda x: tuple(
        (v is None, v) for v in x))
    import torch

    spec_fields = [{"name": (f"{c}_{t}" if t != "identity" else c),
                    "transform": (f"{t}[{p}]" if p is not None else t),
                    "source-id": src_idx[j] + 1, "field-id": 1000 + j}
                   for j, (t, c, p) in enumerate(specs)]
    first = not IcebergTable(path).exists()
    for i, key in enumerate(uniq):
        rows = torch.tensor([r for r, tp in enumerate(tuples) if tp == key],
                            dtype=torch.int64)
        sub = _Chunk([c.gather(rows.to(c.device)) for c in chunk.columns],
                     list(chunk.names))
        pvals = {f["name"]: v for f, v in zip(spec_fields, key)}
        write(path, sub,
              mode if i == 0 else "append", options,
              partition_values=pvals, partition_spec=spec_fields)
    return len(uniq)
