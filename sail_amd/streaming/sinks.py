"""Streaming sinks (ref: sail-data-source console/noop sinks, memory sink,
file/delta sinks through the batch writers)."""
from __future__ import annotations

import os
from typing import Callable, Dict, Optional

from ..engine.chunk import Chunk


class StreamSink:
    def write(self, chunk: Chunk, batch_id: int, output_mode: str):
        raise NotImplementedError


class MemorySink(StreamSink):
    """Registers results as a queryable table in the session catalog
    (complete: replace; append/update: accumulate)."""

    def __init__(self, session, name: str):
        self.session = session
        self.name = name
        self._initialized = False

    def write(self, chunk, batch_id, output_mode):
        cat = self.session.catalog
        if output_mode == "complete" or not self._initialized:
            cat.register_table(self.name, chunk.to_table(),
                               [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)])
            self._initialized = True
        else:
            cat.insert_into(self.name, chunk)


class ConsoleSink(StreamSink):
    def __init__(self, num_rows: int = 20):
        self.num_rows = num_rows

    def write(self, chunk, batch_id, output_mode):
        print(f"-------------------------------------------\n"
              f"Batch: {batch_id}\n"
              f"-------------------------------------------")
        rows = list(zip(*[c.to_pylist()[: self.num_rows] for c in chunk.columns])) \
            if chunk.columns else []
        print(" | ".join(chunk.names))
        for r in rows:
            print(" | ".join(str(v) for v in r))
        if chunk.num_rows > self.num_rows:
            print(f"... ({chunk.num_rows} rows)")


class ForeachBatchSink(StreamSink):
    def __init__(self, session, fn: Callable):
        self.session = session
        self.fn = fn

    def write(self, chunk, batch_id, output_mode):
        self.fn(_BatchView(chunk), batch_id)


class _BatchView:
    """What foreachBatch receives: a tiny DataFrame-like over the batch."""

    def __init__(self, chunk: Chunk):
        self.chunk = chunk
        self.columns = list(chunk.names)

    @property
    def num_rows(self):
        return self.chunk.num_rows

    def to_pydict(self):
        return {n: c.to_pylist() for n, c in zip(self.chunk.names, self.chunk.columns)}

    def collect(self):
        cols = [c.to_pylist() for c in self.chunk.columns]
        return list(zip(*cols)) if cols else []


class FileSink(StreamSink):
    """Append one file set per batch via the batch writers, exactly-once.

    Idempotence (the reference relies on Spark's _spark_metadata /
    delta txn versions for this):
      * delta: each batch commit carries a `txn` action keyed by the query's
        stable id; a replayed batch whose version is already committed is
        skipped (ref: sail-delta-lake transaction app transactions).
      * parquet/csv/json: files are named part-<batch_id>-...; a
        `_sail_metadata/<batch_id>` manifest is O_EXCL-created after the
        data files. On replay, a manifested batch is skipped; an
        unmanifested partial batch has its stray files deleted and is
        rewritten.
    """

    def __init__(self, fmt: str, path: str, options: Dict[str, str]):
        self.fmt = fmt
        self.path = path
        self.options = options or {}
        self.app_id: Optional[str] = None  # set by StreamingQuery (stable id)

    def _manifest_dir(self) -> str:
        d = os.path.join(self.path, "_sail_metadata")
        os.makedirs(d, exist_ok=True)
        return d

    def committed_batch(self, batch_id: int) -> bool:
        if self.fmt == "delta":
            if self.app_id is None:
                return False
            from ..datasource import delta

            try:
                last = delta.last_txn_version(self.path, self.app_id)
            except FileNotFoundError:
                return False
            return last is not None and last >= batch_id
        return os.path.exists(os.path.join(self.path, "_sail_metadata",
                                           str(batch_id)))

    def write(self, chunk, batch_id, output_mode):
        import glob
        import json as _json
        import uuid as _uuid

        from ..datasource.registry import write_source

        if self.committed_batch(batch_id):
            return
        if self.fmt == "delta":
            from ..datasource import delta

            txn = (self.app_id, batch_id) if self.app_id is not None else None
            delta.write(self.path, chunk, "append", self.options, txn=txn)
            return
        # drop stray files from a crashed attempt at this batch
        for stray in glob.glob(os.path.join(self.path,
                                            f"part-{batch_id:05d}-*")):
            os.remove(stray)
        ext = {"parquet": "parquet", "csv": "csv", "json": "json"}[self.fmt]
        os.makedirs(self.path, exist_ok=True)
        fname = f"part-{batch_id:05d}-{_uuid.uuid4().hex}.{ext}"
        target = os.path.join(self.path, fname)
        write_source(self.fmt, target, chunk, "append", self.options, None)
        manifest = os.path.join(self._manifest_dir(), str(batch_id))
        fd = os.open(manifest, os.O_CREAT | os.O_EXCL | os.O_WRONLY)
        with os.fdopen(fd, "w") as f:
            _json.dump({"batchId": batch_id, "files": [fname]}, f)


class NoopSink(StreamSink):
    def write(self, chunk, batch_id, output_mode):
        pass


def make_sink(session, fmt: Optional[str], path: Optional[str],
              options: Dict[str, str], query_name: Optional[str],
              foreach_batch: Optional[Callable]) -> StreamSink:
    if foreach_batch is not None:
        return ForeachBatchSink(session, foreach_batch)
    fmt = (fmt or "memory").lower()
    if fmt == "memory":
        if not query_name:
            raise ValueError("memory sink requires .query_name(...)")
        return MemorySink(session, query_name)
    if fmt == "console":
        return ConsoleSink(int(options.get("numRows", "20")))
    if fmt == "noop":
        return NoopSink()
    if fmt in ("parquet", "csv", "json", "delta"):
        if not path:
            raise ValueError(f"{fmt} sink requires a path")
        return FileSink(fmt, path, options)
    raise ValueError(f"unknown stream sink format: {fmt}")
