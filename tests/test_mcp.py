"""MCP server: JSON-RPC handling and the stdio transport
(ref: sail-cli/src/spark/mcp_server.rs role)."""
import json
import subprocess
import sys

import pytest

from sail_amd.mcp.server import McpServer


@pytest.fixture()
def srv():
    s = McpServer(device="cpu")
    s.session.create_dataframe({"a": [1, 2, 3]}, name="t")
    return s


def _call(srv, method, params=None, mid=1):
    return srv.handle({"jsonrpc": "2.0", "id": mid, "method": method,
                       "params": params or {}})


def test_initialize_and_tools_list(srv):
    r = _call(srv, "initialize")
    assert r["result"]["serverInfo"]["name"] == "sail-mi355x"
    assert srv.handle({"jsonrpc": "2.0", "method": "notifications/initialized"}) is None
    tools = _call(srv, "tools/list")["result"]["tools"]
    assert {t["name"] for t in tools} >= {"run_sql", "list_tables", "describe_table"}


def test_run_sql_tool(srv):
    r = _call(srv, "tools/call",
              {"name": "run_sql", "arguments": {"sql": "SELECT sum(a) AS s FROM t"}})
    out = r["result"]["content"][0]["text"]
    assert not r["result"]["isError"]
    assert out.splitlines() == ["s", "6"]


def test_list_and_describe(srv):
    r = _call(srv, "tools/call", {"name": "list_tables", "arguments": {}})
    assert "t" in r["result"]["content"][0]["text"].split()
    r = _call(srv, "tools/call", {"name": "describe_table", "arguments": {"table": "t"}})
    assert r["result"]["content"][0]["text"].startswith("a\t")


def test_sql_error_is_tool_error(srv):
    r = _call(srv, "tools/call",
              {"name": "run_sql", "arguments": {"sql": "SELECT * FROM missing"}})
    assert r["result"]["isError"]


def test_unknown_method(srv):
    r = _call(srv, "bogus/method")
    assert r["error"]["code"] == -32601


def test_stdio_transport_subprocess(tmp_path):
    msgs = [
        {"jsonrpc": "2.0", "id": 1, "method": "initialize", "params": {}},
        {"jsonrpc": "2.0", "method": "notifications/initialized"},
        {"jsonrpc": "2.0", "id": 2, "method": "tools/call",
         "params": {"name": "run_sql", "arguments": {"sql": "SELECT 1 + 1 AS x"}}},
    ]
    inp = "".join(json.dumps(m) + "\n" for m in msgs)
    proc = subprocess.run(
        [sys.executable, "-m", "sail_amd", "mcp", "server", "--device", "cpu"],
        input=inp, capture_output=True, text=True, timeout=120)
    lines = [json.loads(l) for l in proc.stdout.splitlines() if l.strip()]
    assert lines[0]["id"] == 1
    assert lines[1]["id"] == 2
    assert "2" in lines[1]["result"]["content"][0]["text"]
