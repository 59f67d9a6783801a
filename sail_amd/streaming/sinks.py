"""Streaming sinks (ref: sail-data-source console/noop sinks, memory sink,
file/delta sinks through the batch writers)."""
from __future__ import annotations

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
    """Append one file set per batch via the batch writers."""

    def __init__(self, fmt: str, path: str, options: Dict[str, str]):
        self.fmt = fmt
        self.path = path
        self.options = options or {}

    def write(self, chunk, batch_id, output_mode):
        from ..datasource.registry import write_source

        if self.fmt == "delta":
            from ..datasource import delta

            delta.write(self.path, chunk, "append", self.options)
        else:
            write_source(self.fmt, self.path, chunk, "append", self.options, None)


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
