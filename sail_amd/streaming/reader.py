"""readStream / writeStream API (ref: Spark DataStreamReader/Writer surface;
SURVEY §2.3 write/writeStream commands)."""
from __future__ import annotations

from typing import Callable, Dict, Optional

from .query import StreamingQuery
from .sinks import make_sink
from .sources import StreamSource, make_source

_STREAM_SEQ = [0]


class DataStreamReader:
    def __init__(self, session):
        self._session = session
        self._format = "rate"
        self._options: Dict[str, str] = {}
        self._schema = None

    def format(self, fmt: str) -> "DataStreamReader":
        self._format = fmt
        return self

    def option(self, key: str, value) -> "DataStreamReader":
        self._options[key] = str(value)
        return self

    def options(self, **kw) -> "DataStreamReader":
        for k, v in kw.items():
            self._options[k] = str(v)
        return self

    def schema(self, schema) -> "DataStreamReader":
        """schema: list of (name, DataType), {'name': DataType}, or a DDL
        string like 'k STRING, v INT' (PySpark form)."""
        if isinstance(schema, str):
            from ..sql.parser import parse_ddl_schema

            schema = parse_ddl_schema(schema)
        if isinstance(schema, dict):
            schema = list(schema.items())
        self._schema = schema
        return self

    def load(self, path: Optional[str] = None,
             name: Optional[str] = None) -> "StreamingDataFrame":
        source = make_source(self._format, path, self._options, self._schema)
        if name is None:
            _STREAM_SEQ[0] += 1
            name = f"stream_{_STREAM_SEQ[0]}"
        # session-wide registry: stream-stream joins resolve the OTHER
        # streaming views referenced by a query through this map
        reg = getattr(self._session, "_stream_sources", None)
        if reg is None:
            reg = self._session._stream_sources = {}
        reg[name.lower()] = source
        return StreamingDataFrame(self._session, source, name)


class StreamingDataFrame:
    """A streaming relation: `view_name` is how SQL refers to the incoming
    rows. `.sql(...)` attaches the per-batch transformation."""

    is_streaming = True

    def __init__(self, session, source: StreamSource, view_name: str,
                 query_sql: Optional[str] = None, watermark=None):
        self.session = session
        self.source = source
        self.view_name = view_name
        self.query_sql = query_sql or f"SELECT * FROM {view_name}"
        self.watermark = watermark  # (event_time_col, delay_str)

    def sql(self, query: str) -> "StreamingDataFrame":
        return StreamingDataFrame(self.session, self.source, self.view_name,
                                  query, self.watermark)

    def with_watermark(self, col: str, delay: str) -> "StreamingDataFrame":
        """Declare `col` as the event-time column with max lateness `delay`
        (ref: Spark Dataset.withWatermark). Enables append output mode for
        windowed aggregations: a window is emitted once, when the watermark
        (max event time seen - delay) passes its end."""
        return StreamingDataFrame(self.session, self.source, self.view_name,
                                  self.query_sql, (col, delay))

    withWatermark = with_watermark

    @property
    def write_stream(self) -> "DataStreamWriter":
        return DataStreamWriter(self)

    writeStream = write_stream  # Spark spelling


class DataStreamWriter:
    def __init__(self, sdf: StreamingDataFrame):
        self._sdf = sdf
        self._format: Optional[str] = None
        self._options: Dict[str, str] = {}
        self._output_mode = "append"
        self._query_name: Optional[str] = None
        self._trigger_interval = 0.1
        self._trigger_once = False
        self._available_now = False
        self._foreach_batch: Optional[Callable] = None

    def format(self, fmt: str) -> "DataStreamWriter":
        self._format = fmt
        return self

    def option(self, key: str, value) -> "DataStreamWriter":
        self._options[key] = str(value)
        return self

    def output_mode(self, mode: str) -> "DataStreamWriter":
        mode = mode.lower()
        if mode not in ("append", "complete", "update"):
            raise ValueError(f"unknown output mode {mode}")
        self._output_mode = mode
        return self

    outputMode = output_mode

    def query_name(self, name: str) -> "DataStreamWriter":
        self._query_name = name
        return self

    queryName = query_name

    def trigger(self, processing_time: Optional[float] = None,
                once: bool = False, available_now: bool = False) -> "DataStreamWriter":
        if processing_time is not None:
            self._trigger_interval = float(processing_time)
        self._trigger_once = once
        self._available_now = available_now
        return self

    def foreach_batch(self, fn: Callable) -> "DataStreamWriter":
        self._foreach_batch = fn
        return self

    foreachBatch = foreach_batch

    def toTable(self, tableName: str) -> StreamingQuery:
        """writeStream.toTable(t): memory-table sink named t (Spark
        writes to a catalog table; the engine's tables are in-memory)."""
        self._format = "memory"
        self._query_name = tableName
        return self.start()

    def start(self, path: Optional[str] = None) -> StreamingQuery:
        sdf = self._sdf
        sink = make_sink(sdf.session, self._format, path, self._options,
                         self._query_name, self._foreach_batch)
        q = StreamingQuery(
            sdf.session, sdf.source, sdf.query_sql, sdf.view_name, sink,
            output_mode=self._output_mode,
            trigger_interval=self._trigger_interval,
            trigger_once=self._trigger_once,
            available_now=self._available_now,
            checkpoint_location=self._options.get("checkpointLocation"),
            name=self._query_name, watermark=sdf.watermark)
        sdf.session.streams.register(q)
        return q.start()


class StreamingQueryManager:
    """session.streams (ref: Spark StreamingQueryManager)."""

    def __init__(self):
        self._queries = []

    def register(self, q: StreamingQuery):
        self._queries.append(q)

    @property
    def active(self):
        return [q for q in self._queries if q.is_active]

    def get(self, qid: str) -> Optional[StreamingQuery]:
        for q in self._queries:
            if q.id == qid or q.name == qid:
                return q
        return None

    def stop_all(self):
        for q in self.active:
            q.stop()
