"""Spark Connect client (wire-subset).

A lightweight client speaking the same spark.connect wire subset as
connect/server.py — the pysail-equivalent entry point for this engine
(ref: python/pysail/spark/__init__.py SparkConnectServer usage) and the
conformance harness for the server's wire encoding.
"""
from __future__ import annotations

import io
import uuid
from typing import Dict, List, Optional

import grpc

from . import wire as W
from .server import F, _SERVICE


class ConnectClient:
    def __init__(self, address: str, session_id: Optional[str] = None):
        self._channel = grpc.insecure_channel(address)
        self.session_id = session_id or str(uuid.uuid4())

    def close(self):
        self._channel.close()

    # -- request builders --------------------------------------------------
    def _sql_plan(self, sql: str, command: bool = False) -> bytes:
        sql_msg = W.field_string(F.SQL_QUERY, sql)
        if command:
            return W.field_message(F.PLAN_COMMAND,
                                   W.field_message(F.CMD_SQL,
                                                   W.field_string(F.SQLCMD_SQL, sql)))
        rel = W.field_message(F.REL_SQL, sql_msg)
        return W.field_message(F.PLAN_ROOT, rel)

    # -- RPCs --------------------------------------------------------------
    def sql(self, sql: str, command: bool = False):
        """Execute SQL; returns a pyarrow.Table (possibly empty)."""
        import pyarrow as pa

        req = (W.field_string(F.EXEC_SESSION_ID, self.session_id)
               + W.field_message(F.EXEC_PLAN, self._sql_plan(sql, command))
               + W.field_string(F.EXEC_OPERATION_ID, str(uuid.uuid4())))
        call = self._channel.unary_stream(
            f"/{_SERVICE}/ExecutePlan",
            request_serializer=None, response_deserializer=None)
        batches: List[pa.RecordBatch] = []
        for resp in call(req):
            fields = W.parse(resp)
            ab = W.first(fields, F.RESP_ARROW_BATCH)
            if ab is not None:
                abf = W.parse(ab)
                data = W.first(abf, F.AB_DATA, b"")
                with pa.ipc.open_stream(io.BytesIO(data)) as r:
                    for b in r:
                        batches.append(b)
        if not batches:
            return pa.table({})
        return pa.Table.from_batches(batches)

    def spark_version(self) -> str:
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_SPARK_VERSION, b""))
        resp = self._call_unary("AnalyzePlan", req)
        fields = W.parse(resp)
        ver = W.parse(W.first(fields, F.ANR_SPARK_VERSION, b""))
        return W.first_str(ver, 1)

    def explain(self, sql: str) -> str:
        inner = W.field_message(1, self._sql_plan(sql))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_EXPLAIN, inner))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        ex = W.parse(W.first(fields, F.ANR_EXPLAIN, b""))
        return W.first_str(ex, 1)

    def schema(self, sql: str) -> str:
        inner = W.field_message(1, self._sql_plan(sql))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_SCHEMA, inner))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        sc = W.parse(W.first(fields, F.ANR_SCHEMA, b""))
        return W.first_str(sc, 2)

    def ddl_parse(self, ddl: str) -> str:
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_DDL_PARSE, W.field_string(1, ddl)))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        dp = W.parse(W.first(fields, F.ANR_DDL_PARSE, b""))
        return W.first_str(dp, 2)

    def same_semantics(self, sql_a: str, sql_b: str) -> bool:
        inner = (W.field_message(1, self._sql_plan(sql_a))
                 + W.field_message(2, self._sql_plan(sql_b)))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_SAME_SEMANTICS, inner))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        ss = W.parse(W.first(fields, F.ANR_SAME_SEMANTICS, b""))
        return bool(W.first_varint(ss, 1, 0))

    def semantic_hash(self, sql: str) -> int:
        inner = W.field_message(1, self._sql_plan(sql))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_SEMANTIC_HASH, inner))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        sh = W.parse(W.first(fields, F.ANR_SEMANTIC_HASH, b""))
        return int(W.first_varint(sh, 1, 0))

    def set_conf(self, key: str, value: str):
        kv = W.field_string(F.KV_KEY, key) + W.field_string(F.KV_VALUE, value)
        op = W.field_message(F.CFG_OP_SET, W.field_message(1, kv))
        req = (W.field_string(F.CFG_SESSION_ID, self.session_id)
               + W.field_message(F.CFG_OPERATION, op))
        self._call_unary("Config", req)

    def get_conf(self, key: str) -> Optional[str]:
        op = W.field_message(F.CFG_OP_GET, W.field_string(1, key))
        req = (W.field_string(F.CFG_SESSION_ID, self.session_id)
               + W.field_message(F.CFG_OPERATION, op))
        fields = W.parse(self._call_unary("Config", req))
        for kv in fields.get(F.CFGR_PAIRS, []):
            kvf = W.parse(kv)
            if W.first_str(kvf, F.KV_KEY) == key:
                return W.first_str(kvf, F.KV_VALUE) or None
        return None

    def _call_unary(self, method: str, req: bytes) -> bytes:
        call = self._channel.unary_unary(
            f"/{_SERVICE}/{method}",
            request_serializer=None, response_deserializer=None)
        return call(req)

    def json_to_ddl(self, json_schema: str) -> str:
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_JSON_TO_DDL,
                                 W.field_string(1, json_schema)))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        jd = W.parse(W.first(fields, F.ANR_JSON_TO_DDL, b""))
        return W.first_str(jd, 1)

    def input_files(self, sql: str) -> list:
        inner = W.field_message(1, self._sql_plan(sql))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_INPUT_FILES, inner))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        body = W.parse(W.first(fields, F.ANR_INPUT_FILES, b""))
        return W.all_strs(body, 1)

    def persist(self, sql: str) -> None:
        inner = W.field_message(1, self._sql_plan(sql))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_PERSIST, inner))
        self._call_unary("AnalyzePlan", req)

    def get_storage_level(self, sql: str) -> bool:
        inner = W.field_message(1, self._sql_plan(sql))
        req = (W.field_string(F.AN_SESSION_ID, self.session_id)
               + W.field_message(F.AN_GET_STORAGE_LEVEL, inner))
        fields = W.parse(self._call_unary("AnalyzePlan", req))
        lvl = W.parse(W.first(fields, F.ANR_GET_STORAGE_LEVEL, b""))
        sl = W.parse(W.first(lvl, 1, b""))
        return bool(W.first_varint(sl, 2, 0))
