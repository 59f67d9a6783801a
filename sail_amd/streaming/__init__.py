"""Structured Streaming (micro-batch).

The analogue of the reference's streaming subsystem (ref: SURVEY §5.4;
crates/sail-logical-plan/src/streaming/, sail-physical-plan/src/streaming/,
sail-data-source rate/socket sources + console/noop sinks,
sail-common-datafusion FlowEvent model). The reference threads
Chandy-Lamport flow markers through pull-based operator streams; here the
natural MI355X shape is a micro-batch loop — each trigger materializes the
new rows as a whole-partition device chunk and runs the (already compiled)
batch plan over it, with streaming aggregation kept incremental through the
same partial/merge decomposition the distributed two-phase aggregate uses
(exec/distributed.decompose_agg). Checkpointing is a write-ahead offset log
(offsets/ before the batch, commits/ after the sink) plus a persisted
aggregation-state table, mirroring Spark's checkpoint directory layout.
"""
from .query import StreamingQuery
from .reader import DataStreamReader, DataStreamWriter, StreamingDataFrame
from .sinks import make_sink
from .sources import make_source

__all__ = [
    "StreamingQuery", "DataStreamReader", "DataStreamWriter",
    "StreamingDataFrame", "make_source", "make_sink",
]
