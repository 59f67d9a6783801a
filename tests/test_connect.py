"""Spark Connect server: gRPC round-trips over the wire subset
(ref: crates/sail-spark-connect tests; no PySpark client in this image —
the in-repo client speaks the same wire format)."""
import pytest

import sail_amd
from sail_amd.connect.client import ConnectClient
from sail_amd.connect.server import SparkConnectServer


@pytest.fixture(scope="module")
def server():
    srv = SparkConnectServer(host="127.0.0.1", port=0, device="cpu").start()
    yield srv
    srv.stop()


@pytest.fixture()
def client(server):
    c = ConnectClient(server.address)
    yield c
    c.close()


def test_spark_version(client):
    assert "sail" in client.spark_version()


def test_sql_select(client):
    t = client.sql("SELECT 1 AS a, 'x' AS b")
    assert t.column_names == ["a", "b"]
    assert t.to_pydict() == {"a": [1], "b": ["x"]}


def test_sql_with_data(server, client):
    sess = server.session(client.session_id)
    sess.create_dataframe({"v": [3, 1, 2]}, name="nums")
    t = client.sql("SELECT v * 10 AS x FROM nums ORDER BY v")
    assert t.to_pydict() == {"x": [10, 20, 30]}


def test_command_create_view_and_query(client):
    client.sql("CREATE OR REPLACE TEMP VIEW cv AS SELECT 41 + 1 AS answer", command=True)
    t = client.sql("SELECT answer FROM cv")
    assert t.to_pydict() == {"answer": [42]}


def test_config_roundtrip(client):
    client.set_conf("spark.sql.shuffle.partitions", "7")
    assert client.get_conf("spark.sql.shuffle.partitions") == "7"
    assert client.get_conf("nonexistent.key") is None


def test_explain_and_schema(server, client):
    sess = server.session(client.session_id)
    sess.create_dataframe({"a": [1]}, name="t1")
    assert "Project" in client.explain("SELECT a + 1 FROM t1")
    assert "a" in client.schema("SELECT a FROM t1")


def test_multi_batch_result(server, client):
    sess = server.session(client.session_id)
    sess.create_dataframe({"i": list(range(200_000))}, name="big")
    t = client.sql("SELECT i FROM big")
    assert t.num_rows == 200_000


def test_tpch_q1_over_connect(server, client):
    """BASELINE config #1: TPC-H Q1 at SF small over Spark Connect on CPU."""
    from sail_amd.datagen.tpch import register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES

    sess = server.session(client.session_id)
    register_tpch(sess, sf=0.01)
    t = client.sql(QUERIES[1])
    assert t.num_rows == 4
    assert t.column_names[0] == "l_returnflag"


def test_analyze_ddl_semantics(client):
    out = client.ddl_parse("a INT, b STRING")
    assert "a" in out and "b" in out
    assert client.same_semantics("SELECT 1 + 1", "SELECT 1 + 1")
    assert not client.same_semantics("SELECT 1", "SELECT 2")
    h1 = client.semantic_hash("SELECT 1 + 1")
    h2 = client.semantic_hash("SELECT 1 + 1")
    assert h1 == h2 and h1 > 0


def test_analyze_extended_types(client):
    assert client.json_to_ddl('{"type":"struct","fields":'
                              '[{"name":"a","type":"long"},'
                              '{"name":"b","type":"string"}]}') == "a BIGINT,b STRING"
    assert client.get_storage_level("SELECT 1") is True
    client.persist("SELECT 1")  # no-op ack
    assert client.input_files("SELECT 1") == []


# -- relation-tree (DataFrame API) plans over the wire ----------------------
# (PySpark DataFrame clients send Relation protos, not SQL;
#  ref: crates/sail-spark-connect/src/proto/plan.rs)

from sail_amd.connect.client import E, R


@pytest.fixture()
def rel_data(server, client):
    sess = server.session(client.session_id)
    sess.create_dataframe(
        {"k": ["a", "b", "a", "c", "b", "a"],
         "v": [1, 2, 3, 4, 5, 6],
         "w": [1.5, 2.5, 3.5, 4.5, 5.5, 6.5]}, name="rt")
    sess.create_dataframe(
        {"k": ["a", "b", "z"], "tag": [10, 20, 30]}, name="rt2")
    return sess


def test_relation_project_filter(client, rel_data):
    rel = R.project(
        R.filter(R.read_table("rt"),
                 E.fn(">", E.col("v"), E.lit_long(2))),
        E.col("k"), E.alias(E.fn("*", E.col("v"), E.lit_long(10)), "v10"))
    t = client.execute_relation(rel)
    assert t.column_names == ["k", "v10"]
    assert t.to_pydict() == {"k": ["a", "c", "b", "a"], "v10": [30, 40, 50, 60]}


def test_relation_aggregate_sort(client, rel_data):
    rel = R.sort(
        R.aggregate(R.read_table("rt"),
                    group=[E.col("k")],
                    aggs=[E.alias(E.fn("sum", E.col("v")), "sv"),
                          E.alias(E.fn("count", E.col("v")), "c")]),
        E.sort_order(E.col("k")))
    t = client.execute_relation(rel)
    assert t.to_pydict() == {"k": ["a", "b", "c"], "sv": [10, 7, 4],
                             "c": [3, 2, 1]}


def test_relation_join_limit(client, rel_data):
    rel = R.limit(
        R.sort(
            R.join(R.read_table("rt"), R.read_table("rt2"), how=1,
                   using=["k"]),
            E.sort_order(E.col("v"))),
        3)
    t = client.execute_relation(rel)
    assert t.num_rows == 3
    assert t.to_pydict()["tag"] == [10, 20, 10]


def test_relation_local_and_setop(client):
    import pyarrow as pa

    local = R.local_relation(pa.table({"x": [1, 2, 3]}))
    local2 = R.local_relation(pa.table({"x": [3, 4]}))
    rel = R.sort(R.set_op(local, local2, kind=2, is_all=True),
                 E.sort_order(E.col("x")))
    t = client.execute_relation(rel)
    assert t.to_pydict() == {"x": [1, 2, 3, 3, 4]}


def test_relation_range_withcolumns_drop_rename(client):
    rel = R.range(0, 5)
    rel = R.with_columns(rel, [R.alias_payload(
        E.fn("*", E.col("id"), E.lit_long(2)), "dbl")])
    rel = R.with_columns_renamed(rel, {"dbl": "double_id"})
    rel = R.drop(rel, "id")
    t = client.execute_relation(rel)
    assert t.column_names == ["double_id"]
    assert t.to_pydict() == {"double_id": [0, 2, 4, 6, 8]}


def test_relation_dedup_and_todf(client, rel_data):
    rel = R.to_df(R.deduplicate(R.project(R.read_table("rt"), E.col("k")),
                                all_columns=True), "key")
    t = client.execute_relation(rel)
    assert t.column_names == ["key"]
    assert sorted(t.to_pydict()["key"]) == ["a", "b", "c"]
    # subset dedup keeps one row per key with all columns
    rel2 = R.deduplicate(R.read_table("rt"), columns=["k"])
    t2 = client.execute_relation(rel2)
    assert t2.num_rows == 3 and set(t2.column_names) == {"k", "v", "w"}


def test_relation_show_string_and_tail(client, rel_data):
    rel = R.show_string(R.sort(R.read_table("rt"), E.sort_order(E.col("v"))),
                        num_rows=3)
    t = client.execute_relation(rel)
    text = t.to_pydict()["show_string"][0]
    assert "only showing top 3 rows" in text and "| k" in text
    t2 = client.execute_relation(
        R.tail(R.sort(R.read_table("rt"), E.sort_order(E.col("v"))), 2))
    assert t2.to_pydict()["v"] == [5, 6]


def test_relation_cast_and_distinct_agg(client, rel_data):
    rel = R.aggregate(
        R.read_table("rt"), group=[],
        aggs=[E.alias(E.fn("count", E.col("k"), distinct=True), "dk"),
              E.alias(E.fn("sum", E.cast(E.col("w"), "bigint")), "sw")])
    t = client.execute_relation(rel)
    assert t.to_pydict() == {"dk": [3], "sw": [21]}


def test_reattach_and_release(server, client, rel_data):
    """Dropped result stream -> ReattachExecute replays from the last seen
    response id; ReleaseExecute(until) trims; release_all forgets the op."""
    import uuid as _uuid

    op = str(_uuid.uuid4())
    # big enough for several 65536-row arrow batches
    sess = server.session(client.session_id)
    sess.create_dataframe({"n": list(range(200_000))}, name="big_rt")
    rel = R.read_table("big_rt")
    got = client.execute_relation_raw(rel, op, stop_after=1)
    assert len(got) == 1
    first_rid = got[0][0]
    # reattach after the first response: the rest of the batches arrive
    rest = client.reattach(op, first_rid)
    assert len(rest) >= 1
    all_ids = [r[0] for r in got + rest]
    assert len(all_ids) == len(set(all_ids))  # no duplicates
    total = 0
    import io as _io

    import pyarrow as pa

    from sail_amd.connect import wire as W2
    from sail_amd.connect.server import F as F2

    for _, resp in got + rest:
        fields = W2.parse(resp)
        ab = W2.first(fields, F2.RESP_ARROW_BATCH)
        if ab is not None:
            data = W2.first(W2.parse(ab), F2.AB_DATA, b"")
            with pa.ipc.open_stream(_io.BytesIO(data)) as r:
                for b in r:
                    total += b.num_rows
    assert total == 200_000
    # release everything up to the first id, reattach returns only later ones
    client.release_until(op, first_rid)
    later = client.reattach(op, None)
    assert first_rid not in [r[0] for r in later]
    client.release_all(op)
    import grpc as _grpc

    with pytest.raises(_grpc.RpcError):
        client.reattach(op, None)


def test_add_artifacts_and_status(server, client):
    """AddArtifacts (single-chunk batch + chunked stream) and
    ArtifactStatus (ref: sail-spark-connect server.rs:288-356)."""
    res = client.add_artifact("pyfiles/helper.py", b"def f():\n    return 7\n")
    assert res == [("pyfiles/helper.py", True)]
    client.add_artifact_chunked("jars/big.bin",
                                [b"a" * 100, b"b" * 100, b"c" * 50])
    st = client.artifact_statuses(
        ["pyfiles/helper.py", "jars/big.bin", "missing.txt"])
    assert st == {"pyfiles/helper.py": True, "jars/big.bin": True,
                  "missing.txt": False}
    # stored content is intact on the server side
    arts = server._session_artifacts(client.session_id)
    assert arts["jars/big.bin"] == b"a" * 100 + b"b" * 100 + b"c" * 50


def test_relation_rollup_cube_grouping_sets(client, rel_data):
    """GroupType ROLLUP(2)/CUBE(3)/GROUPING_SETS(5) over the wire."""
    rel = R.sort(
        R.aggregate(R.read_table("rt"), group=[E.col("k")],
                    aggs=[E.alias(E.fn("sum", E.col("v")), "sv")],
                    group_type=2),
        E.sort_order(E.col("k")))
    t = client.execute_relation(rel)
    d = t.to_pydict()
    # rollup adds the grand-total row (k = NULL)
    assert d["sv"] == [21, 10, 7, 4] or d["sv"][0] == 21
    assert sum(1 for k in d["k"] if k is None) == 1

    cube = R.aggregate(R.read_table("rt"), group=[E.col("k")],
                       aggs=[E.alias(E.fn("count", E.col("v")), "c")],
                       group_type=3)
    d2 = client.execute_relation(cube).to_pydict()
    assert sorted(x for x in d2["c"]) == [1, 2, 3, 6]

    gs = R.aggregate(R.read_table("rt"), group=[E.col("k")],
                     aggs=[E.alias(E.fn("sum", E.col("v")), "sv")],
                     group_type=5,
                     grouping_sets=[[E.col("k")], []])
    d3 = client.execute_relation(gs).to_pydict()
    assert sorted(v for v in d3["sv"]) == [4, 7, 10, 21]


def test_relation_pivot(client, server, rel_data):
    sess = server.session(client.session_id)
    sess.create_dataframe(
        {"g": ["x", "x", "y", "y", "y"],
         "k": ["a", "b", "a", "a", "b"],
         "v": [1, 2, 3, 4, 5]}, name="pv_rt")
    rel = R.sort(
        R.aggregate(R.read_table("pv_rt"), group=[E.col("g")],
                    aggs=[E.fn("sum", E.col("v"))],
                    group_type=4, pivot_col=E.col("k"),
                    pivot_values=[E.raw_lit_str("a"), E.raw_lit_str("b")]),
        E.sort_order(E.col("g")))
    d = client.execute_relation(rel).to_pydict()
    assert d["g"] == ["x", "y"]
    assert d["a"] == [1, 7] and d["b"] == [2, 5]
