"""Model Context Protocol server: Spark SQL for LLM agents.

The role of the reference's MCP server (ref: crates/sail-cli/src/spark/
mcp_server.rs:39, run_spark_mcp_server — exposes Spark SQL over MCP so
agent frameworks can query data). Transport here is MCP stdio
(newline-delimited JSON-RPC 2.0); tools execute through the same
SessionContext pipeline as Spark Connect and Flight.
"""
from __future__ import annotations

import json
import sys
from typing import Optional

PROTOCOL_VERSION = "2024-11-05"

TOOLS = [
    {
        "name": "run_sql",
        "description": "Execute a Spark SQL query and return the result rows.",
        "inputSchema": {
            "type": "object",
            "properties": {
                "sql": {"type": "string", "description": "The SQL statement"},
                "limit": {"type": "integer",
                          "description": "Max rows to return (default 100)"},
            },
            "required": ["sql"],
        },
    },
    {
        "name": "list_tables",
        "description": "List the tables and views registered in the session.",
        "inputSchema": {"type": "object", "properties": {}},
    },
    {
        "name": "describe_table",
        "description": "Describe a table's columns and types.",
        "inputSchema": {
            "type": "object",
            "properties": {"table": {"type": "string"}},
            "required": ["table"],
        },
    },
    {
        "name": "read_data",
        "description": "Register a parquet/csv/json/delta path as a table.",
        "inputSchema": {
            "type": "object",
            "properties": {
                "path": {"type": "string"},
                "format": {"type": "string",
                           "enum": ["parquet", "csv", "json", "delta"]},
                "name": {"type": "string", "description": "table name"},
            },
            "required": ["path", "name"],
        },
    },
]


class McpServer:
    """Transport-independent MCP request handler (stdio wrapper below)."""

    def __init__(self, session=None, device: Optional[str] = None):
        if session is None:
            from ..engine.session import SessionContext

            session = SessionContext(device=device)
        self.session = session

    # -- JSON-RPC dispatch --------------------------------------------------
    def handle(self, msg: dict) -> Optional[dict]:
        method = msg.get("method", "")
        mid = msg.get("id")
        try:
            if method == "initialize":
                result = {
                    "protocolVersion": PROTOCOL_VERSION,
                    "capabilities": {"tools": {}},
                    "serverInfo": {"name": "sail-mi355x", "version": "0.1.0"},
                }
            elif method in ("notifications/initialized", "initialized"):
                return None  # notification, no response
            elif method == "ping":
                result = {}
            elif method == "tools/list":
                result = {"tools": TOOLS}
            elif method == "tools/call":
                result = self._call_tool(msg.get("params", {}))
            else:
                if mid is None:
                    return None  # unknown notification
                return _err(mid, -32601, f"method not found: {method}")
        except Exception as e:  # tool errors surface as MCP tool errors
            if method == "tools/call":
                return {"jsonrpc": "2.0", "id": mid, "result": {
                    "content": [{"type": "text", "text": f"error: {e}"}],
                    "isError": True}}
            return _err(mid, -32603, str(e))
        if mid is None:
            return None
        return {"jsonrpc": "2.0", "id": mid, "result": result}

    # -- tools --------------------------------------------------------------
    def _call_tool(self, params: dict) -> dict:
        name = params.get("name")
        args = params.get("arguments") or {}
        if name == "run_sql":
            text = self._run_sql(args["sql"], int(args.get("limit", 100)))
        elif name == "list_tables":
            text = "\n".join(self.session.catalog.list_tables()) or "(no tables)"
        elif name == "describe_table":
            sch = self.session.catalog.table_schema(args["table"])
            if sch is None:
                vp = self.session.catalog.view_plan(args["table"])
                if vp is None:
                    raise ValueError(f"table not found: {args['table']}")
                text = f"{args['table']} is a view"
            else:
                text = "\n".join(f"{n}\t{t!r}" for n, t in sch)
        elif name == "read_data":
            fmt = args.get("format", "parquet")
            df = self.session.read.format(fmt).load(args["path"])
            self.session.catalog.create_view(args["name"], df.plan, replace=True)
            text = f"registered {args['name']} ({fmt}) from {args['path']}"
        else:
            raise ValueError(f"unknown tool {name}")
        return {"content": [{"type": "text", "text": text}], "isError": False}

    def _run_sql(self, sql: str, limit: int) -> str:
        df = self.session.sql(sql)
        chunk = df.collect_chunk()
        names = list(chunk.names)
        cols = [c.to_pylist()[:limit] for c in chunk.columns]
        lines = ["\t".join(names)]
        nrows = min(chunk.num_rows, limit)
        for i in range(nrows):
            lines.append("\t".join(str(c[i]) for c in cols))
        if chunk.num_rows > limit:
            lines.append(f"... ({chunk.num_rows} rows total, showing {limit})")
        return "\n".join(lines)


def _err(mid, code, message):
    return {"jsonrpc": "2.0", "id": mid, "error": {"code": code, "message": message}}


def run_stdio_server(device: Optional[str] = None,
                     stdin=None, stdout=None):
    """Newline-delimited JSON-RPC over stdio (MCP stdio transport)."""
    stdin = stdin or sys.stdin
    stdout = stdout or sys.stdout
    server = McpServer(device=device)
    for line in stdin:
        line = line.strip()
        if not line:
            continue
        try:
            msg = json.loads(line)
        except json.JSONDecodeError:
            continue
        resp = server.handle(msg)
        if resp is not None:
            stdout.write(json.dumps(resp) + "\n")
            stdout.flush()
