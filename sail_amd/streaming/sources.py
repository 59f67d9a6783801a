"""Streaming sources.

Spark's source contract — `latest_offset()` (pure) + `read_between(a, b)`
(replayable) — so the write-ahead offset log in query.py gives effective
exactly-once for replayable sources (ref: sail-data-source rate/socket
sources, delta streaming reads).

Offsets are plain JSON values:
  rate    -> int rows elapsed (time-based, like Spark's rate source)
  memory  -> int rows appended
  file    -> sorted list of ingested file names
  delta   -> int table version
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Optional, Tuple

from ..engine import types as T
from ..engine.chunk import Chunk
from ..engine.column import Column, Table


class StreamSource:
    schema: List[Tuple[str, T.DataType]] = []

    def initial_offset(self):
        raise NotImplementedError

    def latest_offset(self):
        raise NotImplementedError

    def read_between(self, start, end) -> Chunk:
        """Rows with offsets in (start, end]; must be replayable."""
        raise NotImplementedError


class RateSource(StreamSource):
    """Monotonic (timestamp_ms, value) rows at a fixed rate.
    Options: rowsPerSecond (default 1000), numPartitions ignored."""

    def __init__(self, options: Dict[str, str]):
        self.rows_per_second = int(options.get("rowsPerSecond", "1000"))
        self.start_time = time.time()
        self.schema = [("timestamp", T.I64), ("value", T.I64)]

    def initial_offset(self):
        return 0

    def latest_offset(self):
        return int((time.time() - self.start_time) * self.rows_per_second)

    def read_between(self, start, end) -> Chunk:
        import torch

        values = torch.arange(start, end, dtype=torch.int64)
        base_ms = int(self.start_time * 1000)
        ts = base_ms + (values * 1000) // max(self.rows_per_second, 1)
        return Chunk([Column(T.I64, ts), Column(T.I64, values)],
                     ["timestamp", "value"])


class MemorySource(StreamSource):
    """In-process feed for tests and notebooks: `add_rows({...})`."""

    def __init__(self, schema: List[Tuple[str, T.DataType]]):
        self.schema = schema
        self._rows: Dict[str, list] = {n: [] for n, _ in schema}
        self._count = 0
        self._lock = threading.Lock()

    def add_rows(self, data: Dict[str, list]):
        with self._lock:
            n = len(next(iter(data.values())))
            for name, _ in self.schema:
                self._rows[name].extend(data[name])
            self._count += n

    def initial_offset(self):
        return 0

    def latest_offset(self):
        with self._lock:
            return self._count

    def read_between(self, start, end) -> Chunk:
        with self._lock:
            cols = [Column.from_values(self._rows[n][start:end], t)
                    for n, t in self.schema]
        return Chunk(cols, [n for n, _ in self.schema])


class FileSource(StreamSource):
    """Tail a directory of parquet/csv/json files; each new file becomes part
    of the next micro-batch (ref: Spark FileStreamSource semantics)."""

    def __init__(self, fmt: str, path: str, options: Dict[str, str],
                 schema: Optional[List[Tuple[str, T.DataType]]] = None):
        self.fmt = fmt
        self.path = path
        self.options = options
        self._ext = {"parquet": ".parquet", "csv": ".csv", "json": ".json"}[fmt]
        if schema is None:
            files = self._list_files()
            if not files:
                raise ValueError(
                    f"file stream over empty dir {path}: pass .schema(...)")
            from ..datasource.registry import infer_source_schema

            schema = infer_source_schema(fmt, [os.path.join(path, files[0])], options)
        self.schema = schema

    def _list_files(self) -> List[str]:
        if not os.path.isdir(self.path):
            return []
        return sorted(f for f in os.listdir(self.path)
                      if f.endswith(self._ext) and not f.startswith((".", "_")))

    def initial_offset(self):
        return []

    def latest_offset(self):
        return self._list_files()

    def read_between(self, start, end) -> Chunk:
        from ..datasource.registry import read_source

        new = [f for f in end if f not in set(start)]
        if not new:
            return self._empty()
        tbl = read_source(self.fmt, [os.path.join(self.path, f) for f in new],
                          self.options, self.schema, "cpu")
        return Chunk.from_table(tbl)

    def _empty(self) -> Chunk:
        return Chunk([Column.from_values([], t) for _, t in self.schema],
                     [n for n, _ in self.schema])


class DeltaSource(StreamSource):
    """Tail a Delta table's transaction log: offset = version, a batch is the
    add-actions of versions (start, end]. Non-append changes (removes from
    overwrite/MERGE) raise unless option ignoreChanges=true, matching Spark's
    delta streaming contract."""

    def __init__(self, path: str, options: Dict[str, str]):
        from ..datasource.delta import DeltaLog

        self.path = path
        self.options = options
        self.log = DeltaLog(path)
        self.ignore_changes = str(options.get("ignoreChanges", "false")).lower() == "true"
        schema, _, _, _ = self.log.snapshot()
        self.schema = schema

    def initial_offset(self):
        if str(self.options.get("startingVersion", "")).lower() == "earliest":
            return -1
        # default: start from the current snapshot? Spark processes the full
        # table as batch 0; do the same: initial offset = -1 (everything).
        return -1

    def latest_offset(self):
        v = self.log.latest_version()
        return -1 if v is None else v

    def read_between(self, start, end) -> Chunk:
        from ..datasource import parquet_io

        files: List[str] = []
        for v in range(start + 1, end + 1):
            p = os.path.join(self.log.log_path, f"{v:020d}.json")
            with open(p) as f:
                for line in f:
                    if not line.strip():
                        continue
                    action = json.loads(line)
                    if "add" in action:
                        files.append(action["add"]["path"])
                    elif "remove" in action and not self.ignore_changes:
                        raise RuntimeError(
                            f"delta stream source: version {v} removes files "
                            "(overwrite/MERGE); set ignoreChanges=true to skip")
        if not files:
            return Chunk([Column.from_values([], t) for _, t in self.schema],
                         [n for n, _ in self.schema])
        tbl = parquet_io.read([os.path.join(self.path, f) for f in files],
                              self.schema, "cpu", self.options)
        return Chunk.from_table(tbl)


class IcebergSource(StreamSource):
    """Iceberg snapshot-tail source: offset = index into the snapshot log;
    each micro-batch reads the data files ADDED by the new snapshots
    (append snapshots only — overwrites raise unless ignoreChanges)."""

    def __init__(self, path: str, options: Dict[str, str]):
        from ..datasource.iceberg import IcebergTable

        self.path = path
        self.options = options
        self.ignore_changes = str(options.get("ignoreChanges", "false")
                                  ).lower() == "true"
        t = IcebergTable(path)
        if not t.exists():
            raise ValueError(f"not an Iceberg table: {path}")
        self.schema = t.schema()

    def _snapshots(self):
        from ..datasource.iceberg import IcebergTable

        t = IcebergTable(path=self.path)
        return t, (t.metadata.get("snapshots", []) if t.exists() else [])

    def initial_offset(self):
        return -1

    def latest_offset(self):
        _, snaps = self._snapshots()
        return len(snaps) - 1

    def read_between(self, start, end) -> Chunk:
        from ..datasource import parquet_io
        from ..utils.avro import read_container

        t, snaps = self._snapshots()
        files: List[str] = []
        for i in range(start + 1, end + 1):
            snap = snaps[i]
            op = snap.get("summary", {}).get("operation", "append")
            if op != "append" and i > 0 and not self.ignore_changes:
                raise RuntimeError(
                    f"iceberg stream source: snapshot {snap['snapshot-id']} "
                    f"is {op}; set ignoreChanges=true to skip rewrites")
            _, manifests, _ = read_container(t._local(snap["manifest-list"]))
            for mf in manifests:
                if mf.get("added_snapshot_id") != snap["snapshot-id"]:
                    continue  # carried-forward manifest: already emitted
                _, entries, _ = read_container(t._local(mf["manifest_path"]))
                for e in entries:
                    if e.get("status") != 1:  # ADDED in this snapshot
                        continue
                    df = e["data_file"]
                    if df.get("content", 0) == 0:
                        files.append(t._local(df["file_path"]))
        if not files:
            return Chunk([Column.from_values([], ty) for _, ty in self.schema],
                         [n for n, _ in self.schema])
        tbl = parquet_io.read(files, self.schema, "cpu", self.options)
        return Chunk.from_table(tbl)


def make_source(fmt: str, path: Optional[str], options: Dict[str, str],
                schema=None) -> StreamSource:
    fmt = fmt.lower()
    if fmt == "rate":
        return RateSource(options)
    if fmt == "memory":
        if schema is None:
            raise ValueError("memory stream source requires .schema(...)")
        return MemorySource(schema)
    if fmt == "delta":
        return DeltaSource(path, options)
    if fmt == "iceberg":
        return IcebergSource(path, options)
    if fmt in ("parquet", "csv", "json"):
        return FileSource(fmt, path, options, schema)
    raise ValueError(f"unknown stream source format: {fmt}")
