"""Structured streaming: sources, incremental aggregation state, sinks,
offset-WAL checkpoint recovery (ref: SURVEY §5.4 streaming checkpointing)."""
import time

import pytest

import sail_amd
from sail_amd.engine import types as T


@pytest.fixture()
def s():
    return sail_amd.SessionContext(device="cpu")


def _mem_stream(s, name="events"):
    return s.read_stream.format("memory").schema(
        {"k": T.STRING, "v": T.I64}).load(name=name)


def test_stateless_foreach_batch(s):
    sdf = _mem_stream(s)
    src = sdf.source
    seen = []
    q = (sdf.sql("SELECT k, v * 2 AS v2 FROM events WHERE v > 0")
         .write_stream.foreach_batch(lambda df, bid: seen.append((bid, df.to_pydict())))
         .trigger(processing_time=0.01).start())
    src.add_rows({"k": ["a", "b"], "v": [1, -1]})
    q.process_all_available()
    src.add_rows({"k": ["c"], "v": [3]})
    q.process_all_available()
    q.stop()
    assert q.exception is None
    assert seen[0][1] == {"k": ["a"], "v2": [2]}
    assert seen[1][1] == {"k": ["c"], "v2": [6]}


def test_incremental_aggregation_complete(s):
    sdf = _mem_stream(s)
    src = sdf.source
    q = (sdf.sql("SELECT k, sum(v) AS sv, count(*) AS n FROM events GROUP BY k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("agg_out").trigger(processing_time=0.01).start())
    assert q._mode == "incremental"
    src.add_rows({"k": ["a", "b", "a"], "v": [1, 2, 3]})
    q.process_all_available()
    rows = dict((r[0], (r[1], r[2])) for r in s.sql(
        "SELECT k, sv, n FROM agg_out").collect())
    assert rows == {"a": (4, 2), "b": (2, 1)}
    src.add_rows({"k": ["b", "c"], "v": [10, 5]})
    q.process_all_available()
    rows = dict((r[0], (r[1], r[2])) for r in s.sql(
        "SELECT k, sv, n FROM agg_out").collect())
    assert rows == {"a": (4, 2), "b": (12, 2), "c": (5, 1)}
    q.stop()


def test_update_mode_emits_touched_only(s):
    sdf = _mem_stream(s)
    src = sdf.source
    batches = []
    q = (sdf.sql("SELECT k, sum(v) AS sv FROM events GROUP BY k")
         .write_stream.output_mode("update")
         .foreach_batch(lambda df, bid: batches.append(df.to_pydict()))
         .trigger(processing_time=0.01).start())
    src.add_rows({"k": ["a", "b"], "v": [1, 2]})
    q.process_all_available()
    src.add_rows({"k": ["b"], "v": [5]})
    q.process_all_available()
    q.stop()
    assert sorted(batches[0]["k"]) == ["a", "b"]
    assert batches[1] == {"k": ["b"], "sv": [7]}  # only the touched group


def test_avg_and_upper_plan(s):
    """avg decomposes to sum+count; ORDER BY above the aggregate runs on the
    merged state through the ChunkSource splice."""
    sdf = _mem_stream(s)
    src = sdf.source
    q = (sdf.sql("SELECT k, avg(v) AS m FROM events GROUP BY k ORDER BY k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("avg_out").trigger(processing_time=0.01).start())
    assert q._mode == "incremental"
    src.add_rows({"k": ["a", "a"], "v": [1, 2]})
    q.process_all_available()
    src.add_rows({"k": ["a", "b"], "v": [6, 4]})
    q.process_all_available()
    q.stop()
    assert s.sql("SELECT k, m FROM avg_out ORDER BY k").collect() == [
        ("a", 3.0), ("b", 4.0)]


def test_retained_fallback_distinct(s):
    sdf = _mem_stream(s)
    src = sdf.source
    q = (sdf.sql("SELECT count(DISTINCT k) AS d FROM events")
         .write_stream.output_mode("complete").format("memory")
         .query_name("d_out").trigger(processing_time=0.01).start())
    assert q._mode == "retained"
    src.add_rows({"k": ["a", "b"], "v": [1, 1]})
    q.process_all_available()
    src.add_rows({"k": ["a", "c"], "v": [1, 1]})
    q.process_all_available()
    q.stop()
    assert s.sql("SELECT d FROM d_out").collect() == [(3,)]


def test_rate_source(s):
    q = (s.read_stream.format("rate").option("rowsPerSecond", 5000)
         .load(name="ticks")
         .sql("SELECT count(*) AS n, max(value) AS mx FROM ticks")
         .write_stream.output_mode("complete").format("memory")
         .query_name("rate_out").trigger(processing_time=0.01).start())
    deadline = time.time() + 5
    n = 0
    while time.time() < deadline:
        rows = s.sql("SELECT n, mx FROM rate_out").collect() \
            if s.catalog.table_schema("rate_out") else []
        if rows and rows[0][0] and rows[0][0] > 100:
            n, mx = rows[0]
            break
        time.sleep(0.02)
    q.stop()
    assert q.exception is None
    assert n > 100 and mx == n - 1  # values are 0..n-1 exactly once


def test_file_source_available_now(s, tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    d = tmp_path / "in"
    d.mkdir()
    pq.write_table(pa.table({"x": [1, 2, 3]}), d / "a.parquet")
    sdf = s.read_stream.format("parquet").load(str(d), name="files")
    q = (sdf.sql("SELECT sum(x) AS sx FROM files")
         .write_stream.output_mode("complete").format("memory")
         .query_name("f_out").trigger(available_now=True).start())
    q.await_termination(timeout=20)
    assert s.sql("SELECT sx FROM f_out").collect() == [(6,)]
    # new file, second availableNow pass on the same query object state
    pq.write_table(pa.table({"x": [10]}), d / "b.parquet")
    q2 = (sdf.sql("SELECT sum(x) AS sx FROM files")
          .write_stream.output_mode("complete").format("memory")
          .query_name("f_out2").trigger(available_now=True).start())
    q2.await_termination(timeout=20)
    assert s.sql("SELECT sx FROM f_out2").collect() == [(16,)]


def test_delta_source_and_sink(s, tmp_path):
    from sail_amd.datasource import delta

    src_path = str(tmp_path / "dsrc")
    out_path = str(tmp_path / "dout")
    s.create_dataframe({"k": ["a", "b"], "v": [1, 2]},
                       schema={"k": T.STRING, "v": T.I64}, name="seed")
    s.table("seed").write.format("delta").mode("overwrite").save(src_path)
    sdf = s.read_stream.format("delta").load(src_path, name="dtail")
    q = (sdf.sql("SELECT k, v FROM dtail")
         .write_stream.format("delta").trigger(processing_time=0.01)
         .start(out_path))
    q.process_all_available()
    # append a new version to the source table; the stream should tail it
    s.create_dataframe({"k": ["c"], "v": [9]},
                       schema={"k": T.STRING, "v": T.I64}, name="more")
    s.table("more").write.format("delta").mode("append").save(src_path)
    q.process_all_available()
    q.stop()
    assert q.exception is None
    rows = s.sql(f"SELECT k, v FROM delta.`{out_path}` ORDER BY k").collect()
    assert rows == [("a", 1), ("b", 2), ("c", 9)]


def test_checkpoint_recovery(s, tmp_path):
    ckpt = str(tmp_path / "ckpt")
    schema = {"k": T.STRING, "v": T.I64}
    sdf = s.read_stream.format("memory").schema(schema).load(name="ev1")
    src = sdf.source
    q = (sdf.sql("SELECT k, sum(v) AS sv FROM ev1 GROUP BY k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("c_out").option("checkpointLocation", ckpt)
         .trigger(processing_time=0.01).start())
    src.add_rows({"k": ["a", "b"], "v": [1, 2]})
    q.process_all_available()
    q.stop()
    # "restart": a fresh session + query over the same checkpoint; the memory
    # source starts empty but the aggregation state is restored from ckpt.
    s2 = sail_amd.SessionContext(device="cpu")
    sdf2 = s2.read_stream.format("memory").schema(schema).load(name="ev1")
    src2 = sdf2.source
    # replay what the recovered offset says was already committed
    src2.add_rows({"k": ["a", "b"], "v": [1, 2]})
    q2 = (sdf2.sql("SELECT k, sum(v) AS sv FROM ev1 GROUP BY k")
          .write_stream.output_mode("complete").format("memory")
          .query_name("c_out").option("checkpointLocation", ckpt)
          .trigger(processing_time=0.01).start())
    assert q2.batch_id == q.batch_id  # offsets resumed, batch not re-run
    src2.add_rows({"k": ["a"], "v": [10]})
    q2.process_all_available()
    q2.stop()
    assert q2.exception is None
    rows = dict(s2.sql("SELECT k, sv FROM c_out").collect())
    assert rows == {"a": 11, "b": 2}  # state restored: 1+10, not 10


def test_append_with_aggregation_rejected(s):
    sdf = _mem_stream(s, name="ev2")
    with pytest.raises(ValueError):
        sdf.sql("SELECT k, sum(v) FROM ev2 GROUP BY k") \
           .write_stream.output_mode("append").format("noop").start()


HOUR_US = 3_600_000_000
MIN_US = 60_000_000


def _wm_query(s, name, out, mode="append", ckpt=None):
    r = s.read_stream.format("memory").schema([("ts", T.TIMESTAMP), ("v", T.I64)])
    sdf = r.load(name=name)
    src = sdf.source
    sdf = sdf.sql(f"SELECT window(ts, '1 hour').start AS ws, sum(v) AS sv "
                  f"FROM {name} GROUP BY window(ts, '1 hour')"
                  ).with_watermark("ts", "30 minutes")
    w = sdf.write_stream.format("memory").query_name(out).output_mode(mode) \
        .trigger(processing_time=0.01)
    if ckpt:
        w = w.option("checkpointLocation", ckpt)
    return src, w.start()


def test_watermark_append_mode(s):
    src, q = _wm_query(s, "wm_ev1", "wm_out1")
    try:
        src.add_rows({"ts": [10 * MIN_US, 50 * MIN_US], "v": [1, 2]})
        q.process_all_available()
        assert q.watermark_us == 20 * MIN_US  # max(50m) - 30m
        # nothing closed yet -> no output table
        src.add_rows({"ts": [2 * HOUR_US + 40 * MIN_US], "v": [5]})
        q.process_all_available()
        # watermark 2h10m closes the hour-0 window
        assert s.sql("SELECT * FROM wm_out1 ORDER BY 1").collect() == [(0, 3)]
        # late row (5m < watermark) must be DROPPED; 3h50m closes hour-2
        src.add_rows({"ts": [5 * MIN_US, 3 * HOUR_US + 50 * MIN_US], "v": [100, 7]})
        q.process_all_available()
        assert s.sql("SELECT * FROM wm_out1 ORDER BY 1").collect() == \
            [(0, 3), (2 * HOUR_US, 5)]
    finally:
        q.stop()


def test_watermark_update_mode_evicts_state(s):
    src, q = _wm_query(s, "wm_ev2", "wm_out2", mode="update")
    try:
        src.add_rows({"ts": [10 * MIN_US], "v": [1]})
        q.process_all_available()
        src.add_rows({"ts": [3 * HOUR_US], "v": [2]})
        q.process_all_available()
        # hour-0 window evicted once watermark passed; state holds 1 group
        assert len(q._agg_state.keys[0]) == 1
    finally:
        q.stop()


def test_watermark_append_requires_time_key(s):
    r = s.read_stream.format("memory").schema([("k", T.I64), ("v", T.I64)])
    sdf = r.load(name="wm_ev3").sql("SELECT k, sum(v) FROM wm_ev3 GROUP BY k")
    with pytest.raises(ValueError, match="append output mode"):
        sdf.write_stream.format("memory").query_name("wm_out3") \
            .output_mode("append").start()


def test_watermark_checkpoint_recovery(s, tmp_path):
    ck = str(tmp_path / "wm_ck")
    src, q = _wm_query(s, "wm_ev4", "wm_out4", ckpt=ck)
    try:
        src.add_rows({"ts": [10 * MIN_US, HOUR_US + 10 * MIN_US], "v": [1, 2]})
        q.process_all_available()
    finally:
        q.stop()
    # restart from the checkpoint: watermark and window state survive
    src2, q2 = _wm_query(s, "wm_ev5", "wm_out4", ckpt=ck)
    try:
        assert q2._max_event_us == HOUR_US + 10 * MIN_US
        assert len(q2._agg_state.keys[0]) == 2  # both open windows restored
        src2.add_rows({"ts": [3 * HOUR_US], "v": [9]})
        q2.process_all_available()
        rows = s.sql("SELECT * FROM wm_out4 ORDER BY 1").collect()
        assert (0, 1) in rows and (HOUR_US, 2) in rows
    finally:
        q2.stop()


def test_stream_static_join(s):
    """Stateless streaming plans can join against static catalog tables
    (ref: Spark stream-static joins)."""
    s.create_dataframe({"k": [1, 2], "label": ["one", "two"]}, name="dims")
    r = s.read_stream.format("memory").schema([("k", T.I64), ("v", T.I64)])
    sdf = r.load(name="ss_ev")
    src = sdf.source
    q = sdf.sql("SELECT d.label, e.v FROM ss_ev e JOIN dims d ON e.k = d.k") \
        .write_stream.format("memory").query_name("ss_out") \
        .trigger(processing_time=0.01).start()
    try:
        src.add_rows({"k": [1, 2, 1], "v": [10, 20, 30]})
        q.process_all_available()
        rows = sorted(s.sql("SELECT * FROM ss_out").collect())
        assert rows == [("one", 10), ("one", 30), ("two", 20)]
    finally:
        q.stop()


def test_iceberg_stream_source(s, tmp_path):
    base = str(tmp_path / "ice_stream")
    s.create_dataframe({"id": [1, 2]}, name="ist_src")
    s.sql(f"CREATE TABLE iceberg.`{base}` AS SELECT * FROM ist_src").collect()
    sdf = s.read_stream.format("iceberg").load(base)
    q = sdf.sql(f"SELECT count(*) AS n FROM {sdf.view_name}") \
        .write_stream.format("memory").query_name("ist_out") \
        .output_mode("complete").trigger(processing_time=0.02).start()
    try:
        q.process_all_available()
        assert s.sql("SELECT * FROM ist_out").collect() == [(2,)]
        # appended snapshot arrives as an incremental micro-batch
        s.sql(f"INSERT INTO iceberg.`{base}` VALUES (3)").collect()
        q.process_all_available()
        assert s.sql("SELECT * FROM ist_out").collect() == [(3,)]
    finally:
        q.stop()


def test_retained_mode_checkpoint_recovery(s, tmp_path):
    """retained-mode (non-incremental) input must be checkpointed
    so a restart does not silently drop prior rows while offsets advance."""
    ckpt = str(tmp_path / "ret_ckpt")
    schema = {"k": T.STRING, "v": T.I64}
    sdf = s.read_stream.format("memory").schema(schema).load(name="rv1")
    src = sdf.source
    q = (sdf.sql("SELECT count(DISTINCT k) AS d FROM rv1")
         .write_stream.output_mode("complete").format("memory")
         .query_name("r_out").option("checkpointLocation", ckpt)
         .trigger(processing_time=0.01).start())
    assert q._mode == "retained"
    src.add_rows({"k": ["a", "b"], "v": [1, 1]})
    q.process_all_available()
    q.stop()
    s2 = sail_amd.SessionContext(device="cpu")
    sdf2 = s2.read_stream.format("memory").schema(schema).load(name="rv1")
    src2 = sdf2.source
    src2.add_rows({"k": ["a", "b"], "v": [1, 1]})  # replay committed offsets
    q2 = (sdf2.sql("SELECT count(DISTINCT k) AS d FROM rv1")
          .write_stream.output_mode("complete").format("memory")
          .query_name("r_out").option("checkpointLocation", ckpt)
          .trigger(processing_time=0.01).start())
    assert q2.batch_id == q.batch_id
    src2.add_rows({"k": ["a", "c"], "v": [1, 1]})  # only c is new
    q2.process_all_available()
    q2.stop()
    assert q2.exception is None
    # a,b retained from before the restart: distinct = {a,b,c} = 3
    assert s2.sql("SELECT d FROM r_out").collect() == [(3,)]


def test_state_crash_between_save_and_commit_no_double_count(s, tmp_path):
    """a crash after _save_state but before the commit marker must
    replay the pending batch against the PREVIOUS state (versioned
    snapshots), not against state that already contains it."""
    import json as _json
    import os
    import shutil

    ckpt = str(tmp_path / "dc_ckpt")
    schema = {"k": T.STRING, "v": T.I64}
    sdf = s.read_stream.format("memory").schema(schema).load(name="dc1")
    src = sdf.source
    q = (sdf.sql("SELECT k, sum(v) AS sv FROM dc1 GROUP BY k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("dc_out").option("checkpointLocation", ckpt)
         .trigger(processing_time=0.01).start())
    src.add_rows({"k": ["a"], "v": [1]})
    q.process_all_available()
    q.stop()
    last = q.batch_id
    # forge the crash window: state for batch last+1 exists (as if merged),
    # offsets/last+1 exists, but commits/last+1 does NOT
    nxt = last + 1
    shutil.copytree(os.path.join(ckpt, "state", str(last)),
                    os.path.join(ckpt, "state", str(nxt)))
    # make the forged pending-state distinguishable: as if the pending batch
    # (v=10) was already merged into it (a -> 11)
    import pyarrow as _pa
    import pyarrow.parquet as _pq

    forged = os.path.join(ckpt, "state", str(nxt), "data",
                          "part-00000.parquet")
    _pq.write_table(_pa.table({"k0": ["a"], "p0": [11]}), forged)
    with open(os.path.join(ckpt, "offsets", str(nxt)), "w") as f:
        _json.dump({"offset": 2}, f)
    s2 = sail_amd.SessionContext(device="cpu")
    sdf2 = s2.read_stream.format("memory").schema(schema).load(name="dc1")
    src2 = sdf2.source
    src2.add_rows({"k": ["a"], "v": [1]})   # replayed committed batch
    src2.add_rows({"k": ["a"], "v": [10]})  # the pending batch's rows
    q2 = (sdf2.sql("SELECT k, sum(v) AS sv FROM dc1 GROUP BY k")
          .write_stream.output_mode("complete").format("memory")
          .query_name("dc_out").option("checkpointLocation", ckpt)
          .trigger(processing_time=0.01).start())
    q2.process_all_available()
    q2.stop()
    assert q2.exception is None
    # exactly-once: 1 + 10 (not 1 + 10 + 10 from the forged state)
    assert dict(s2.sql("SELECT k, sv FROM dc_out").collect()) == {"a": 11}


def test_file_sink_replay_idempotent(s, tmp_path):
    """replaying a pending batch after a crash between sink.write
    and the commit marker must not duplicate rows (manifest idempotence)."""
    import os

    ckpt = str(tmp_path / "fs_ckpt")
    out = str(tmp_path / "fs_out")
    schema = {"k": T.STRING, "v": T.I64}
    sdf = s.read_stream.format("memory").schema(schema).load(name="fs1")
    src = sdf.source
    q = (sdf.sql("SELECT k, v FROM fs1")
         .write_stream.format("parquet").option("checkpointLocation", ckpt)
         .trigger(processing_time=0.01).start(out))
    src.add_rows({"k": ["a", "b"], "v": [1, 2]})
    q.process_all_available()
    q.stop()
    last = q.batch_id
    # forge the crash: drop the commit marker; sink files + manifest remain
    os.remove(os.path.join(ckpt, "commits", str(last)))
    s2 = sail_amd.SessionContext(device="cpu")
    sdf2 = s2.read_stream.format("memory").schema(schema).load(name="fs1")
    src2 = sdf2.source
    src2.add_rows({"k": ["a", "b"], "v": [1, 2]})  # pending batch replays
    q2 = (sdf2.sql("SELECT k, v FROM fs1")
          .write_stream.format("parquet").option("checkpointLocation", ckpt)
          .trigger(processing_time=0.01).start(out))
    q2.process_all_available()
    q2.stop()
    assert q2.exception is None
    rows = s2.sql(f"SELECT k, v FROM parquet.`{out}` ORDER BY k").collect()
    assert rows == [("a", 1), ("b", 2)]


def test_delta_sink_replay_idempotent(s, tmp_path):
    """Delta sink idempotence via txn actions keyed by the stable query id."""
    import os

    ckpt = str(tmp_path / "ds_ckpt")
    out = str(tmp_path / "ds_out")
    schema = {"k": T.STRING, "v": T.I64}
    sdf = s.read_stream.format("memory").schema(schema).load(name="ds1")
    src = sdf.source
    q = (sdf.sql("SELECT k, v FROM ds1")
         .write_stream.format("delta").option("checkpointLocation", ckpt)
         .trigger(processing_time=0.01).start(out))
    src.add_rows({"k": ["x"], "v": [7]})
    q.process_all_available()
    q.stop()
    os.remove(os.path.join(ckpt, "commits", str(q.batch_id)))
    s2 = sail_amd.SessionContext(device="cpu")
    sdf2 = s2.read_stream.format("memory").schema(schema).load(name="ds1")
    src2 = sdf2.source
    src2.add_rows({"k": ["x"], "v": [7]})
    q2 = (sdf2.sql("SELECT k, v FROM ds1")
          .write_stream.format("delta").option("checkpointLocation", ckpt)
          .trigger(processing_time=0.01).start(out))
    q2.process_all_available()
    q2.stop()
    assert q2.exception is None
    assert s2.sql(f"SELECT k, v FROM delta.`{out}`").collect() == [("x", 7)]


def test_stream_stream_inner_join(s):
    """Stream-stream join: rows match across batches in both directions
    (ref: Spark stream-stream joins; the reference's streaming rewriter)."""
    left = s.read_stream.format("memory").schema(
        {"k": T.STRING, "lv": T.I64}).load(name="ssl")
    right = s.read_stream.format("memory").schema(
        {"k": T.STRING, "rv": T.I64}).load(name="ssr")
    srcL, srcR = left.source, right.source
    q = (left.sql("SELECT ssl.k, lv, rv FROM ssl JOIN ssr ON ssl.k = ssr.k")
         .write_stream.output_mode("append").format("memory")
         .query_name("ss_out").trigger(processing_time=0.01).start())
    assert q._mode == "multi_retained"
    srcL.add_rows({"k": ["a", "b"], "lv": [1, 2]})
    q.process_all_available()
    # no matches yet
    srcR.add_rows({"k": ["b", "c"], "rv": [20, 30]})
    q.process_all_available()
    # late-arriving left row matches EARLIER right row
    srcL.add_rows({"k": ["c"], "lv": [3]})
    q.process_all_available()
    q.stop()
    assert q.exception is None
    rows = sorted(s.sql("SELECT k, lv, rv FROM ss_out").collect())
    assert rows == [("b", 2, 20), ("c", 3, 30)]


def test_stream_stream_join_complete_and_agg(s):
    left = s.read_stream.format("memory").schema(
        {"k": T.STRING, "lv": T.I64}).load(name="cl")
    right = s.read_stream.format("memory").schema(
        {"k": T.STRING, "rv": T.I64}).load(name="cr")
    q = (left.sql("SELECT cl.k AS k, sum(lv + rv) AS s "
                  "FROM cl JOIN cr ON cl.k = cr.k GROUP BY cl.k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("cc_out").trigger(processing_time=0.01).start())
    left.source.add_rows({"k": ["x", "x"], "lv": [1, 2]})
    right.source.add_rows({"k": ["x"], "rv": [10]})
    q.process_all_available()
    assert dict(s.sql("SELECT k, s FROM cc_out").collect()) == {"x": 23}
    right.source.add_rows({"k": ["x"], "rv": [100]})
    q.process_all_available()
    q.stop()
    assert q.exception is None
    assert dict(s.sql("SELECT k, s FROM cc_out").collect()) == {"x": 226}


def test_stream_stream_join_watermark_evicts_state(s):
    """Retained join state is bounded: rows older than the watermark are
    evicted and can no longer match."""
    left = s.read_stream.format("memory").schema(
        {"ts": T.TIMESTAMP, "k": T.STRING}).load(name="wl")
    right = s.read_stream.format("memory").schema(
        {"ts": T.TIMESTAMP, "k": T.STRING}).load(name="wr")
    hour = 3_600_000_000
    q = (left.with_watermark("ts", "1 hour")
         .sql("SELECT wl.k FROM wl JOIN wr ON wl.k = wr.k")
         .write_stream.output_mode("append").format("memory")
         .query_name("wm_out").trigger(processing_time=0.01).start())
    left.source.add_rows({"ts": [1 * hour], "k": ["old"]})
    q.process_all_available()
    # advance event time far beyond the watermark horizon
    left.source.add_rows({"ts": [10 * hour], "k": ["new"]})
    q.process_all_available()
    # "old" (at 1h) is beyond the 1h watermark of max(10h) -> evicted;
    # a matching right row must NOT join it anymore
    right.source.add_rows({"ts": [10 * hour], "k": ["old"]})
    right.source.add_rows({"ts": [10 * hour], "k": ["new"]})
    q.process_all_available()
    q.stop()
    assert q.exception is None
    rows = s.sql("SELECT k FROM wm_out").collect()
    assert rows == [("new",)]


def test_session_window_batch_and_stream(s):
    """session_window(ts, gap): gap-separated sessions per sibling group
    key, (min_ts, max_ts+gap) structs (ref: Spark session windows)."""
    m = 60_000_000
    s.create_dataframe(
        {"u": ["a", "a", "a", "b", "a", "b"],
         "ts": [0 * m, 3 * m, 20 * m, 1 * m, 22 * m, 2 * m],
         "v": [1, 2, 3, 4, 5, 6]},
        schema={"u": T.STRING, "ts": T.TIMESTAMP, "v": T.I64}, name="sess_ev")
    r = s.sql("SELECT u, session_window(ts, '5 minutes') AS w, sum(v) AS sv "
              "FROM sess_ev GROUP BY u, session_window(ts, '5 minutes')").collect()
    got = sorted((u, w["start"], w["end"], sv) for u, w, sv in r)
    assert got == [
        ("a", 0, 8 * m, 3),            # rows at 0 and 3min merge
        ("a", 20 * m, 27 * m, 8),      # rows at 20 and 22min merge
        ("b", 1 * m, 7 * m, 10),       # rows at 1 and 2min merge
    ]
    # streaming (retained mode re-evaluates sessions each batch)
    sdf = s.read_stream.format("memory").schema(
        {"u": T.STRING, "ts": T.TIMESTAMP, "v": T.I64}).load(name="sw_in")
    src = sdf.source
    q = (sdf.sql("SELECT u, session_window(ts, '5 minutes') AS w, sum(v) s "
                 "FROM sw_in GROUP BY u, session_window(ts, '5 minutes')")
         .write_stream.output_mode("complete").format("memory")
         .query_name("sw_out").trigger(processing_time=0.01).start())
    src.add_rows({"u": ["x"], "ts": [0], "v": [1]})
    q.process_all_available()
    src.add_rows({"u": ["x"], "ts": [2 * m], "v": [10]})  # extends session
    q.process_all_available()
    q.stop()
    assert q.exception is None
    rows = s.sql("SELECT u, w, s FROM sw_out").collect()
    assert [(u, w["start"], w["end"], sv) for u, w, sv in rows] == \
        [("x", 0, 7 * m, 11)]


def test_to_table_and_ddl_schema(s):
    """writeStream.toTable + DDL-string schemas (PySpark forms)."""
    src = s.read_stream.format("memory").schema("k STRING, v INT") \
        .load(name="tt_in")
    q = src.sql("SELECT k, sum(v) sv FROM tt_in GROUP BY k") \
        .write_stream.output_mode("complete").toTable("tt_out")
    src.source.add_rows({"k": ["a", "b", "a"], "v": [1, 2, 3]})
    q.process_all_available()
    assert s.sql("SELECT * FROM tt_out ORDER BY k").collect() == \
        [("a", 4), ("b", 2)]
    q.stop()
