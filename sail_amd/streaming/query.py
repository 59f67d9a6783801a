"""Micro-batch streaming query runner.

One StreamingQuery = one source + one batch plan + one sink, driven by a
trigger loop on a daemon thread:

  1. end = source.latest_offset(); nothing new -> idle (or finish for
     trigger once / availableNow)
  2. WAL: checkpoint offsets/<batch>.json BEFORE executing (Spark layout)
  3. register the batch rows as the stream's table; execute the plan
       - stateless plan  -> plan over just the new rows (append mode)
       - aggregation     -> incremental: per-batch partial aggregates merged
         into a persistent state table via the same partial/merge
         decomposition as the distributed two-phase aggregate
         (exec/distributed.decompose_agg); non-decomposable aggregates
         (DISTINCT, percentiles) fall back to retained-input re-aggregation
  4. sink.write(result, batch_id, mode)
  5. checkpoint commits/<batch>.json + state/ parquet -> batch is durable;
     on restart the query replays any offset without a commit.

The reference threads Chandy-Lamport flow markers through operator streams
(ref: sail-common-datafusion/src/streaming/event/marker.rs FlowMarker
{Watermark, Checkpoint, EndOfData}); micro-batch boundaries play the same
role here — every batch boundary IS a consistent cut, so the offset WAL +
state snapshot gives the same recovery guarantee without per-operator
marker plumbing.
"""
from __future__ import annotations

import json
import os
import threading
import time
import uuid
from typing import List, Optional

import torch

from ..engine import types as T
from ..engine.chunk import Chunk
from ..engine.column import Column
from ..plan import spec as S
from .sinks import StreamSink
from .sources import StreamSource

_PASSTHROUGH = (S.Project, S.Filter, S.Sort, S.Limit, S.SubqueryAlias)


def _find_aggregate(plan: S.Plan):
    """Locate a single Aggregate reachable through passthrough nodes from the
    root. Returns (agg, parent) — parent is None when the root IS the
    aggregate — or None when the plan has no such spine (stateless, or a
    shape the incremental path doesn't cover)."""
    node, parent = plan, None
    while True:
        if isinstance(node, S.Aggregate):
            return node, parent
        if isinstance(node, _PASSTHROUGH):
            parent, node = node, node.input
            continue
        return None


def _count_aggregates(plan: S.Plan) -> int:
    n = 1 if isinstance(plan, S.Aggregate) else 0
    for c in plan.children():
        if c is not None:
            n += _count_aggregates(c)
    return n


class _AggState:
    """Persistent streaming-aggregation state: group keys + partial columns,
    merged with each batch's local partials exactly like the distributed
    merge phase (engine/executor.py _dist_aggregate steps 1+3)."""

    def __init__(self, agg: S.Aggregate, decomps):
        self.agg = agg
        self.decomps = decomps
        self.keys: Optional[List[Column]] = None      # one per group_by
        self.partials: Optional[List[Column]] = None  # flattened per decomp

    # -- persistence --------------------------------------------------------
    def state_chunk(self) -> Optional[Chunk]:
        """State as a flat chunk; struct keys (window()) are flattened to
        one column per field so the parquet state files stay flat."""
        from ..engine.column import StructColumn

        if self.partials is None:
            return None
        cols, names = [], []
        for i, k in enumerate(self.keys or []):
            if isinstance(k, StructColumn):
                for fname, fcol in k.children_:
                    cols.append(Column(fcol.dtype, fcol.data, k.validity))
                    names.append(f"k{i}.{fname}")
            else:
                cols.append(k)
                names.append(f"k{i}")
        cols += list(self.partials)
        names += [f"p{i}" for i in range(len(self.partials))]
        return Chunk(cols, names)

    def load_chunk(self, chunk: Chunk):
        from ..engine.column import StructColumn

        keys, ci = [], 0
        for g in self.agg.group_by:
            dt = getattr(g, "dtype", None)
            if isinstance(dt, T.StructType):
                fields = []
                for f in dt.fields:
                    fields.append((f.name, chunk.columns[ci]))
                    ci += 1
                validity = fields[0][1].validity
                keys.append(StructColumn(fields, validity, dtype=dt))
            else:
                keys.append(chunk.columns[ci])
                ci += 1
        self.keys = keys
        self.partials = list(chunk.columns[ci:])

    # -- update -------------------------------------------------------------
    def update(self, ex, child: Chunk):
        """Merge one batch into the state; returns (finalized_chunk,
        touched_mask over output groups)."""
        from ..engine.aggregates import agg_eval, global_ids, group_ids
        from ..engine.eval import broadcast
        from ..engine.executor import _empty_partial, concat_columns

        agg = self.agg
        n, dev = child.num_rows, child.device
        # 1) batch-local partials
        if agg.group_by:
            key_cols = [broadcast(ex.ev.eval(g, child), n, dev) for g in agg.group_by]
            if n:
                gid, rep, ng = group_ids(key_cols)
                batch_keys = [c.gather(rep) for c in key_cols]
            else:
                gid, ng = torch.zeros(0, dtype=torch.int64, device=dev), 0
                batch_keys = key_cols
        else:
            gid, ng = (global_ids(n, dev) if n
                       else (torch.zeros(0, dtype=torch.int64, device=dev), 0))
            batch_keys = []
        batch_partials: List[Column] = []
        merge_names: List[str] = []
        for a, d in zip(agg.aggs, self.decomps):
            args = [broadcast(ex.ev.eval(x, child), n, dev) for x in a.args]
            fmask = ex.ev.eval_mask(a.filter, child) if a.filter is not None else None
            for pname, mname in zip(d.partials, d.merges):
                use_args = args if (args or pname != "count") else []
                col = agg_eval(pname, use_args, gid, ng, False, fmask,
                               None) if ng else _empty_partial(pname, dev)
                batch_partials.append(col)
                merge_names.append(mname)
        # 2) concat with state (state rows first), re-group, merge
        if self.partials is None:
            all_keys = batch_keys
            all_partials = batch_partials
            n_state = 0
        else:
            all_keys = [concat_columns([s, b]) for s, b in zip(self.keys, batch_keys)]
            all_partials = [concat_columns([s, b])
                            for s, b in zip(self.partials, batch_partials)]
            n_state = self.partials[0].data.shape[0] if self.partials else 0
        total = all_partials[0].data.shape[0] if all_partials else 0
        if agg.group_by:
            mgid, mrep, mng = group_ids(all_keys)
            out_keys = [c.gather(mrep) for c in all_keys]
        else:
            mgid, mng = global_ids(total, dev) if total else (
                torch.zeros(0, dtype=torch.int64, device=dev), 0)
            out_keys = []
        merged = [agg_eval(mname, [col], mgid, mng, False, None, None)
                  for mname, col in zip(merge_names, all_partials)]
        touched = torch.zeros(mng, dtype=torch.bool, device=dev)
        if total > n_state:
            touched[mgid[n_state:]] = True
        # 3) new state + finalized output
        self.keys = out_keys
        self.partials = merged
        out_cols = []
        ci = 0
        for a, d in zip(agg.aggs, self.decomps):
            k = len(d.partials)
            out_cols.append(d.finalize(merged[ci:ci + k], a.dtype))
            ci += k
        out = Chunk(out_keys + out_cols, [nm for nm, _ in agg.schema])
        return out, touched


class StreamingQuery:
    """Handle on a running streaming query (ref: Spark StreamingQuery API:
    stop/awaitTermination/processAllAvailable/lastProgress)."""

    def __init__(self, session, source: StreamSource, sql: str, view_name: str,
                 sink: StreamSink, output_mode: str = "append",
                 trigger_interval: float = 0.1, trigger_once: bool = False,
                 available_now: bool = False,
                 checkpoint_location: Optional[str] = None,
                 name: Optional[str] = None, watermark=None):
        self.id = str(uuid.uuid4())
        self.name = name
        self.session = session
        self.source = source
        self.sql = sql
        self.view_name = view_name
        self.sink = sink
        self.output_mode = output_mode
        self.trigger_interval = trigger_interval
        self.trigger_once = trigger_once
        self.available_now = available_now
        self.checkpoint = checkpoint_location
        self.watermark_spec = watermark          # (event_time_col, delay_str)
        self._wm_delay_us = 0
        self._max_event_us: Optional[int] = None  # max event time ever seen
        self._wm_col_idx: Optional[int] = None    # index into agg.input schema
        self._time_key_idx: Optional[int] = None  # index into agg.group_by
        self.exception: Optional[BaseException] = None
        self.last_progress: Optional[dict] = None
        self.batch_id = -1
        self._offset = source.initial_offset()
        self._stop = threading.Event()
        self._idle = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()
        self._plan = None
        self._agg_state: Optional[_AggState] = None
        self._upper_parent = None
        self._retained: Optional[Chunk] = None  # non-incremental fallback
        #: stream-stream mode: every OTHER streaming view referenced by the
        #: SQL (ref: Spark stream-stream joins; the reference rewrites these
        #: through its StreamingRewriter — here the micro-batch runner
        #: retains both inputs and re-evaluates, emitting deltas)
        self._extra_sources: Dict[str, StreamSource] = {}
        self._multi_retained: Dict[str, Optional[Chunk]] = {}
        self._extra_offsets: Dict[str, object] = {}
        self._prev_result_keys = None
        if self.checkpoint:
            self.id = self._load_or_create_metadata()
        if hasattr(self.sink, "app_id"):
            # stable identity for sink-side idempotence (delta txn /
            # _sail_metadata manifests survive restarts with the query id)
            self.sink.app_id = self.id
        self._prepare()
        if self.checkpoint:
            self._recover()

    # -- planning -----------------------------------------------------------
    def _prepare(self):
        # register an empty table of the source schema so the SQL resolves
        empty = Chunk([Column.from_values([], t) for _, t in self.source.schema],
                      [n for n, _ in self.source.schema])
        self.session.catalog.register_table(
            self.view_name, empty.to_table(), list(self.source.schema))
        reg = getattr(self.session, "_stream_sources", {}) or {}
        refs = set()
        raw = self.session.parse(self.sql)

        def _walk_reads(pl):
            if isinstance(pl, S.Read):
                refs.add(pl.table.lower())
            for ch in (pl.children() if hasattr(pl, "children") else []):
                if ch is not None:
                    _walk_reads(ch)

        _walk_reads(raw)
        for v in sorted(refs):
            if v in reg and v != self.view_name.lower():
                self._extra_sources[v] = reg[v]
                self._multi_retained[v] = None
                self._extra_offsets[v] = reg[v].initial_offset()
                # the extra view must resolve: register its empty schema
                from ..engine.column import Column as _C

                empty2 = Chunk([_C.from_values([], t)
                                for _, t in reg[v].schema],
                               [n for n, _ in reg[v].schema])
                self.session.catalog.register_table(
                    v, empty2.to_table(), list(reg[v].schema))
        self._plan = self.session.plan_sql(self.sql)
        if isinstance(self._plan, S.Command):
            raise ValueError("streaming query must be a SELECT")
        if self._extra_sources:
            self._mode = "multi_retained"
            if self.watermark_spec is not None:
                from ..engine.functions_impl import _parse_duration_us

                self._wm_delay_us = _parse_duration_us(self.watermark_spec[1])
            return
        found = _find_aggregate(self._plan)
        self._mode = "stateless"
        if found is not None and _count_aggregates(self._plan) == 1:
            agg, parent = found
            from ..exec.distributed import decompose_agg

            decomps = [decompose_agg(a) for a in agg.aggs]
            from ..engine.executor import Executor as _Ex

            ok = (not agg.grouping_sets and agg.having is None
                  and all(d is not None for d in decomps)
                  and all(not getattr(a, "distinct", False) for a in agg.aggs)
                  # session windows re-session retroactively as batches
                  # arrive: retained re-evaluation, not incremental merge
                  and _Ex._find_session_window(agg) is None)
            if ok:
                self._mode = "incremental"
                self._agg_state = _AggState(agg, decomps)
                self._upper_parent = parent
        elif _count_aggregates(self._plan) > 0:
            self._mode = "retained"
        if found is not None and self._mode == "stateless":
            self._mode = "retained"
        if self._mode == "stateless" and self.output_mode == "complete":
            self._mode = "retained"  # complete over a stateless plan: re-run all
        if self._mode == "incremental" and self.watermark_spec is not None:
            self._prepare_watermark()
        if self._mode != "stateless" and self.output_mode == "append":
            if self._time_key_idx is None:
                raise ValueError(
                    "append output mode with streaming aggregation requires "
                    "withWatermark() on the event-time column and grouping by "
                    "window(<event-time>, ...) or the event-time column itself")

    def _prepare_watermark(self):
        """Resolve the watermark column against the aggregate's input and
        find which group key carries event time (a window() struct or a
        timestamp-typed key)."""
        from ..engine.functions_impl import _parse_duration_us

        col, delay = self.watermark_spec
        self._wm_delay_us = _parse_duration_us(delay)
        agg = self._agg_state.agg
        in_names = [n for n, _ in agg.input.schema]
        if col not in in_names:
            raise ValueError(
                f"watermark column {col!r} not found below the aggregation "
                f"(have {in_names})")
        self._wm_col_idx = in_names.index(col)
        wdt = agg.input.schema[self._wm_col_idx][1]
        if not (isinstance(wdt, (T.TimestampType, T.DateType)) or wdt.is_integer):
            raise ValueError(
                f"watermark column {col!r} must be timestamp/date/integer, "
                f"got {wdt}")
        for i, g in enumerate(agg.group_by):
            e = g.child if isinstance(g, S.Alias) else g
            if isinstance(e, S.Func) and e.name.lower() == "window":
                self._time_key_idx = i
                break
        else:
            for i, g in enumerate(agg.group_by):
                e = g.child if isinstance(g, S.Alias) else g
                if isinstance(getattr(e, "dtype", None),
                              (T.TimestampType, T.DateType)):
                    self._time_key_idx = i
                    break

    # -- checkpoint ---------------------------------------------------------
    def _ckpt_dir(self, sub: str) -> str:
        d = os.path.join(self.checkpoint, sub)
        os.makedirs(d, exist_ok=True)
        return d

    def _load_or_create_metadata(self) -> str:
        """Stable query id across restarts (Spark checkpoint `metadata`
        layout: {"id": ...})."""
        os.makedirs(self.checkpoint, exist_ok=True)
        p = os.path.join(self.checkpoint, "metadata")
        if os.path.exists(p):
            with open(p) as f:
                return json.load(f)["id"]
        with open(p, "w") as f:
            json.dump({"id": self.id}, f)
        return self.id

    def _recover(self):
        offs = self._ckpt_dir("offsets")
        commits = self._ckpt_dir("commits")
        done = sorted(int(f) for f in os.listdir(commits) if f.isdigit())
        pending = sorted(int(f) for f in os.listdir(offs) if f.isdigit())
        if done:
            last = done[-1]
            with open(os.path.join(offs, str(last))) as f:
                rec = json.load(f)
            self._offset = rec["offset"]
            for v, o in (rec.get("extra") or {}).items():
                if v in self._extra_offsets:
                    self._extra_offsets[v] = o
            self.batch_id = last
            with open(os.path.join(commits, str(last))) as f:
                self._max_event_us = json.load(f).get("maxEventTimeUs")
            # restore state from the snapshot matching the last COMMITTED
            # batch (snapshots are versioned per batch id, so a crash
            # between state write and commit cannot double-count: the
            # pending batch replays against the previous state)
            state_root = os.path.join(self.checkpoint, "state")
            vdir = os.path.join(state_root, str(last))
            if os.path.isdir(vdir):
                self._load_state(vdir)
            elif os.path.isdir(state_root):
                self._load_state(state_root)  # legacy unversioned layout
        # a pending offset without a commit is replayed by the normal loop:
        # read_between(self._offset, that offset) reproduces the batch.
        if pending and (not done or pending[-1] > done[-1]):
            with open(os.path.join(offs, str(pending[-1]))) as f:
                rec = json.load(f)
            self._pending_offset = rec["offset"]
            self._pending_extra = rec.get("extra") or {}
            self._pending_id = pending[-1]
        else:
            self._pending_offset = None

    @staticmethod
    def _load_offset(d, i):
        with open(os.path.join(d, str(i))) as f:
            return json.load(f)["offset"]

    def _save_state(self, bid: int):
        """Versioned per-batch state snapshot under state/<bid>/ — written
        BEFORE the commit marker. Covers incremental aggregation state and
        retained-mode accumulated input (both must survive restart for
        exactly-once results)."""
        if not self.checkpoint:
            return
        kind = "agg"
        chunk = None
        if self._agg_state is not None:
            chunk = self._agg_state.state_chunk()
        elif self._mode == "retained":
            kind, chunk = "retained", self._retained
        if chunk is None:
            return
        from ..datasource.delta import schema_to_string
        from ..datasource.registry import write_source

        state_dir = os.path.join(self._ckpt_dir("state"), str(bid))
        os.makedirs(state_dir, exist_ok=True)
        schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
        write_source("parquet", os.path.join(state_dir, "data"), chunk,
                     "overwrite", {}, None)
        with open(os.path.join(state_dir, "schema.json"), "w") as f:
            json.dump({"kind": kind, "schema": schema_to_string(schema)}, f)

    def _prune_state(self, bid: int):
        """Drop state snapshots older than the previous committed batch."""
        import shutil

        sd = os.path.join(self.checkpoint, "state")
        if not os.path.isdir(sd):
            return
        for f in os.listdir(sd):
            if f.isdigit() and int(f) < bid - 1:
                shutil.rmtree(os.path.join(sd, f), ignore_errors=True)

    def _load_state(self, state_dir):
        from ..datasource.delta import schema_from_string
        from ..datasource.registry import read_source

        sp = os.path.join(state_dir, "schema.json")
        dp = os.path.join(state_dir, "data")
        if not (os.path.exists(sp) and os.path.isdir(dp)):
            return
        with open(sp) as f:
            obj = json.load(f)
        if isinstance(obj, dict) and "kind" in obj:
            kind, schema = obj["kind"], schema_from_string(obj["schema"])
        else:  # legacy: the file IS the schema string
            kind, schema = "agg", schema_from_string(json.dumps(obj))
        files = [os.path.join(dp, f) for f in sorted(os.listdir(dp))
                 if f.endswith(".parquet")]
        tbl = read_source("parquet", files, {}, schema, self.session.device)
        chunk = Chunk.from_table(tbl)
        if kind == "retained":
            self._retained = chunk
        elif self._agg_state is not None:
            self._agg_state.load_chunk(chunk)

    # -- lifecycle ----------------------------------------------------------
    def start(self) -> "StreamingQuery":
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name=f"stream-{self.name or self.id[:8]}")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=30)

    @property
    def is_active(self) -> bool:
        return self._thread is not None and self._thread.is_alive()

    def await_termination(self, timeout: Optional[float] = None) -> bool:
        if self._thread is None:
            return True
        self._thread.join(timeout)
        if self.exception is not None:
            raise self.exception
        return not self._thread.is_alive()

    def process_all_available(self, timeout: float = 30.0):
        """Block until everything currently available has been committed."""
        deadline = time.time() + timeout
        while time.time() < deadline:
            if self.exception is not None:
                raise self.exception
            if not self.is_active:
                return
            if self._offset == self.source.latest_offset() \
                    and all(self._extra_offsets[v] == s2.latest_offset()
                            for v, s2 in self._extra_sources.items()) \
                    and self._idle.is_set():
                return
            time.sleep(0.01)
        raise TimeoutError("process_all_available timed out")

    # -- the loop -----------------------------------------------------------
    def _run(self):
        try:
            while not self._stop.is_set():
                progressed = self._run_one_batch()
                if not progressed:
                    self._idle.set()
                    if self.trigger_once or self.available_now:
                        return
                    self._stop.wait(self.trigger_interval)
                else:
                    self._idle.clear()
                    if self.trigger_once:
                        return
        except BaseException as e:  # surfaced via await_termination
            self.exception = e
        finally:
            self._idle.set()

    def _run_one_batch(self) -> bool:
        if getattr(self, "_pending_offset", None) is not None:
            end, extra_ends, bid = (self._pending_offset,
                                    getattr(self, "_pending_extra", {}),
                                    self._pending_id)
            self._pending_offset = None
        else:
            end = self.source.latest_offset()
            extra_ends = {v: src.latest_offset()
                          for v, src in self._extra_sources.items()}
            if end == self._offset and all(
                    extra_ends[v] == self._extra_offsets[v]
                    for v in extra_ends):
                return False
            bid = self.batch_id + 1
            if self.checkpoint:
                with open(os.path.join(self._ckpt_dir("offsets"), str(bid)), "w") as f:
                    json.dump({"offset": end, "extra": extra_ends}, f)
        t0 = time.time()
        batch = self.source.read_between(self._offset, end)
        nrows = batch.num_rows
        extra_batches = {}
        for v, src in self._extra_sources.items():
            extra_batches[v] = src.read_between(self._extra_offsets[v],
                                                extra_ends[v])
            nrows += extra_batches[v].num_rows
        if self._mode == "multi_retained":
            result = self._execute_multi(batch, extra_batches)
        else:
            result = self._execute_batch(batch)
        if result is not None:
            self.sink.write(result, bid, self.output_mode)
        if self.checkpoint:
            self._save_state(bid)
            with open(os.path.join(self._ckpt_dir("commits"), str(bid)), "w") as f:
                json.dump({"batchId": bid,
                           "maxEventTimeUs": self._max_event_us}, f)
            self._prune_state(bid)
        self._offset = end
        self._extra_offsets.update(extra_ends)
        self.batch_id = bid
        self.last_progress = {
            "id": self.id, "name": self.name, "batchId": bid,
            "numInputRows": nrows,
            "durationMs": round((time.time() - t0) * 1000, 3),
            "sources": [{"endOffset": end}],
        }
        return True

    # -- watermark ----------------------------------------------------------
    @property
    def watermark_us(self) -> Optional[int]:
        """Current watermark in epoch micros (None before any event)."""
        if self._max_event_us is None:
            return None
        return self._max_event_us - self._wm_delay_us

    @staticmethod
    def _to_us(col: Column) -> torch.Tensor:
        us = col.data.to(torch.int64)
        if isinstance(col.dtype, T.DateType):
            us = us * 86_400_000_000
        return us

    def _filter_late(self, child: Chunk) -> Chunk:
        """Drop rows older than the watermark (and null event times)."""
        wm = self.watermark_us
        c = child.columns[self._wm_col_idx]
        keep = self._to_us(c) >= wm if wm is not None else torch.ones(
            len(c), dtype=torch.bool, device=c.device)
        if c.validity is not None:
            keep &= c.validity
        if bool(keep.all()):
            return child
        idx = torch.nonzero(keep, as_tuple=False).flatten()
        return Chunk([col.gather(idx) for col in child.columns],
                     list(child.names))

    def _advance_watermark(self, child: Chunk):
        c = child.columns[self._wm_col_idx]
        us = self._to_us(c)
        if c.validity is not None:
            us = us[c.validity]
        if us.numel():
            m = int(us.max())
            self._max_event_us = m if self._max_event_us is None \
                else max(self._max_event_us, m)

    def _group_close_us(self) -> Optional[torch.Tensor]:
        """Per-state-group close time: window.end for window() keys, the key
        value itself for plain timestamp keys."""
        st = self._agg_state
        if st.keys is None or self._time_key_idx is None:
            return None
        key = st.keys[self._time_key_idx]
        from ..engine.column import StructColumn

        if isinstance(key, StructColumn):
            return self._to_us(dict(key.children_)["end"])
        return self._to_us(key)

    def _expired_mask(self) -> Optional[torch.Tensor]:
        wm = self.watermark_us
        close = self._group_close_us()
        if wm is None or close is None:
            return None
        return close <= wm

    def _evict(self, expired: Optional[torch.Tensor]):
        if expired is None or not bool(expired.any()):
            return
        st = self._agg_state
        keep = torch.nonzero(~expired, as_tuple=False).flatten()
        st.keys = [c.gather(keep) for c in st.keys]
        st.partials = [c.gather(keep) for c in st.partials]

    def _accumulate(self, prev: Optional[Chunk], batch: Chunk) -> Chunk:
        from ..engine.executor import concat_columns

        if prev is None or prev.num_rows == 0:
            return batch
        if batch.num_rows == 0:
            return prev
        return Chunk([concat_columns([a, b]) for a, b in
                      zip(prev.columns, batch.columns)], list(batch.names))

    def _execute_multi(self, batch: Chunk, extra: dict) -> Optional[Chunk]:
        """Stream-stream mode (joins and any other multi-source shape):
        retain every source's input, re-evaluate the whole plan, and for
        append/update emit only rows NEW since the previous evaluation
        (multiset delta on row keys). Watermarks evict retained rows older
        than the event-time horizon."""
        cat = self.session.catalog
        self._retained = self._accumulate(self._retained, batch)
        cat.register_table(self.view_name, self._retained.to_table(),
                           list(self.source.schema))
        for v, b in extra.items():
            self._multi_retained[v] = self._accumulate(
                self._multi_retained.get(v), b)
            cat.register_table(v, self._multi_retained[v].to_table(),
                               list(self._extra_sources[v].schema))
        self._evict_retained_by_watermark()
        result = self.session.execute_plan(self._plan)
        if self.output_mode == "complete":
            return result
        from ..engine.executor import ExecutionContext, Executor

        ex = Executor(ExecutionContext(self.session, self.session.device))
        keys = ex._row_keys(result).cpu().tolist() if result.num_rows else []
        prev = self._prev_result_keys or {}
        counts: dict = {}
        emit_rows = []
        for i, k in enumerate(keys):
            counts[k] = counts.get(k, 0) + 1
            if counts[k] > prev.get(k, 0):
                emit_rows.append(i)
        self._prev_result_keys = counts
        if not emit_rows:
            return None
        idx = torch.tensor(emit_rows, dtype=torch.int64,
                           device=result.device)
        return Chunk([c.gather(idx) for c in result.columns],
                     list(result.names))

    def _evict_retained_by_watermark(self):
        """Drop retained rows older than the watermark on every source that
        carries the event-time column (bounds stream-stream join state)."""
        if self.watermark_spec is None:
            return
        col, _ = self.watermark_spec
        # advance max event time from the primary retained input
        for name, chunk in [(self.view_name, self._retained)] +                 list(self._multi_retained.items()):
            if chunk is None:
                continue
            names_l = [n.lower() for n in chunk.names]
            if col.lower() not in names_l:
                continue
            c = chunk.columns[names_l.index(col.lower())]
            if chunk.num_rows:
                mx = int(c.data.max().item())
                self._max_event_us = max(self._max_event_us or mx, mx)
        wm = self.watermark_us
        if wm is None:
            return
        for name in [self.view_name] + list(self._multi_retained.keys()):
            chunk = self._retained if name == self.view_name                 else self._multi_retained[name]
            if chunk is None:
                continue
            names_l = [n.lower() for n in chunk.names]
            if col.lower() not in names_l:
                continue
            c = chunk.columns[names_l.index(col.lower())]
            keep = c.data >= wm
            if bool(keep.all()):
                continue
            idx = torch.nonzero(keep, as_tuple=False).flatten()
            pruned = Chunk([cc.gather(idx) for cc in chunk.columns],
                           list(chunk.names))
            if name == self.view_name:
                self._retained = pruned
            else:
                self._multi_retained[name] = pruned

    def _execute_batch(self, batch: Chunk) -> Optional[Chunk]:
        from ..engine.executor import ExecutionContext, Executor, concat_columns

        cat = self.session.catalog
        if self._mode == "retained":
            if self._retained is None:
                self._retained = batch
            else:
                self._retained = Chunk(
                    [concat_columns([a, b]) for a, b in
                     zip(self._retained.columns, batch.columns)],
                    list(batch.names))
            cat.register_table(self.view_name, self._retained.to_table(),
                               list(self.source.schema))
            return self.session.execute_plan(self._plan)
        cat.register_table(self.view_name, batch.to_table(),
                           list(self.source.schema))
        if self._mode == "stateless":
            if batch.num_rows == 0:
                return None
            return self.session.execute_plan(self._plan)
        # incremental aggregation
        ctx = ExecutionContext(self.session, self.session.device)
        ex = Executor(ctx)
        child = ex.execute(self._agg_state.agg.input)
        if self._wm_col_idx is not None:
            child = self._filter_late(child)
        finalized, touched = self._agg_state.update(ex, child)
        expired = None
        if self._wm_col_idx is not None:
            self._advance_watermark(child)
            expired = self._expired_mask()
        if self.output_mode == "append":
            # emit only windows the watermark has closed, then drop their state
            idx = torch.nonzero(expired, as_tuple=False).flatten() \
                if expired is not None else torch.zeros(0, dtype=torch.int64)
            out = finalized.gather(idx) if idx.numel() else None
            self._evict(expired)
            if out is None:
                return None
            finalized = out
        elif self.output_mode == "update":
            idx = torch.nonzero(touched, as_tuple=False).flatten()
            finalized = finalized.gather(idx)
            self._evict(expired)
        # complete mode keeps all state (Spark: watermark does not evict there)
        if self._upper_parent is None:
            return finalized
        sub = S.ChunkSource(chunk=finalized, schema=self._agg_state.agg.schema)
        orig = self._upper_parent.input
        try:
            self._upper_parent.input = sub
            return ex.execute(self._plan)
        finally:
            self._upper_parent.input = orig
