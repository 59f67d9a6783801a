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
