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


# ===========================================================================
# Relation-tree (DataFrame API) client surface: builders producing the same
# spark/connect Relation & Expression protos a PySpark client sends
# (relations.proto / expressions.proto field numbers), plus reattach/release.
# ===========================================================================

class E:
    """Expression proto builders (bytes)."""

    @staticmethod
    def col(name: str) -> bytes:
        return W.field_message(2, W.field_string(1, name))

    @staticmethod
    def lit_long(v: int) -> bytes:
        return W.field_message(1, W.field_varint(7, v))

    @staticmethod
    def lit_int(v: int) -> bytes:
        return W.field_message(1, W.field_varint(6, v))

    @staticmethod
    def lit_str(s: str) -> bytes:
        return W.field_message(1, W.field_string(13, s))

    @staticmethod
    def raw_lit_str(s: str) -> bytes:
        """Bare Literal message (pivot values are Literal, not Expression)."""
        return W.field_string(13, s)

    @staticmethod
    def raw_lit_long(v: int) -> bytes:
        return W.field_varint(7, v)

    @staticmethod
    def lit_double(x: float) -> bytes:
        import struct

        raw = struct.unpack("<Q", struct.pack("<d", x))[0]
        return W.field_message(1, W.field_varint(11, raw))

    @staticmethod
    def fn(name: str, *args: bytes, distinct: bool = False) -> bytes:
        body = W.field_string(1, name)
        for a in args:
            body += W.field_message(2, a)
        if distinct:
            body += W.field_varint(3, 1)
        return W.field_message(3, body)

    @staticmethod
    def alias(child: bytes, name: str) -> bytes:
        return W.field_message(6, W.field_message(1, child)
                               + W.field_string(2, name))

    @staticmethod
    def star() -> bytes:
        return W.field_message(5, b"")

    @staticmethod
    def cast(child: bytes, type_str: str) -> bytes:
        return W.field_message(7, W.field_message(1, child)
                               + W.field_string(3, type_str))

    @staticmethod
    def sort_order(child: bytes, asc: bool = True,
                   nulls_first: Optional[bool] = None) -> bytes:
        body = W.field_message(1, child) + W.field_varint(2, 1 if asc else 2)
        if nulls_first is not None:
            body += W.field_varint(3, 1 if nulls_first else 2)
        return body


class R:
    """Relation proto builders (bytes)."""

    @staticmethod
    def read_table(name: str) -> bytes:
        nt = W.field_message(1, W.field_string(1, name))
        return W.field_message(2, nt)

    @staticmethod
    def read_source(fmt: str, paths, options=None) -> bytes:
        ds = W.field_string(1, fmt)
        for p in paths:
            ds += W.field_string(4, p)
        for k, v in (options or {}).items():
            ds += W.field_message(3, W.field_string(1, k) + W.field_string(2, v))
        return W.field_message(2, W.field_message(2, ds))

    @staticmethod
    def project(input_rel: bytes, *exprs: bytes) -> bytes:
        body = W.field_message(1, input_rel)
        for e in exprs:
            body += W.field_message(3, e)
        return W.field_message(3, body)

    @staticmethod
    def filter(input_rel: bytes, condition: bytes) -> bytes:
        return W.field_message(4, W.field_message(1, input_rel)
                               + W.field_message(2, condition))

    @staticmethod
    def join(left: bytes, right: bytes, how: int = 1,
             condition: Optional[bytes] = None, using=None) -> bytes:
        body = W.field_message(1, left) + W.field_message(2, right)
        if condition is not None:
            body += W.field_message(3, condition)
        body += W.field_varint(4, how)
        for c in (using or []):
            body += W.field_string(5, c)
        return W.field_message(5, body)

    @staticmethod
    def aggregate(input_rel: bytes, group: list, aggs: list,
                  group_type: int = 1, pivot_col: bytes = None,
                  pivot_values: list = None,
                  grouping_sets: list = None) -> bytes:
        """group_type: 1=GROUPBY 2=ROLLUP 3=CUBE 4=PIVOT 5=GROUPING_SETS."""
        body = W.field_message(1, input_rel) + W.field_varint(2, group_type)
        for g in group:
            body += W.field_message(3, g)
        for a in aggs:
            body += W.field_message(4, a)
        if group_type == 4 and pivot_col is not None:
            pv = W.field_message(1, pivot_col)
            for lit in pivot_values or []:
                pv += W.field_message(2, lit)
            body += W.field_message(5, pv)
        for gs in grouping_sets or []:
            gsb = b"".join(W.field_message(1, e) for e in gs)
            body += W.field_message(6, gsb)
        return W.field_message(9, body)

    @staticmethod
    def sort(input_rel: bytes, *orders: bytes) -> bytes:
        body = W.field_message(1, input_rel)
        for o in orders:
            body += W.field_message(2, o)
        return W.field_message(7, body)

    @staticmethod
    def limit(input_rel: bytes, n: int) -> bytes:
        return W.field_message(8, W.field_message(1, input_rel)
                               + W.field_varint(2, n))

    @staticmethod
    def set_op(left: bytes, right: bytes, kind: int, is_all: bool) -> bytes:
        body = (W.field_message(1, left) + W.field_message(2, right)
                + W.field_varint(3, kind) + W.field_varint(4, 1 if is_all else 0))
        return W.field_message(6, body)

    @staticmethod
    def local_relation(table) -> bytes:
        """pyarrow.Table -> LocalRelation (Arrow IPC bytes)."""
        import io as _io

        import pyarrow as pa

        sink = _io.BytesIO()
        with pa.ipc.new_stream(sink, table.schema) as w:
            for b in table.to_batches():
                w.write_batch(b)
        return W.field_message(11, W.field_bytes(1, sink.getvalue()))

    @staticmethod
    def range(start: int, end: int, step: int = 1) -> bytes:
        body = (W.field_varint(1, start) + W.field_varint(2, end)
                + W.field_varint(3, step))
        return W.field_message(15, body)

    @staticmethod
    def deduplicate(input_rel: bytes, columns=None, all_columns=False) -> bytes:
        body = W.field_message(1, input_rel)
        for c in (columns or []):
            body += W.field_string(2, c)
        if all_columns:
            body += W.field_varint(3, 1)
        return W.field_message(14, body)

    @staticmethod
    def with_columns(input_rel: bytes, aliases: list) -> bytes:
        body = W.field_message(1, input_rel)
        for al in aliases:
            # aliases are Expression.Alias payloads (not wrapped Expression)
            body += W.field_message(2, al)
        return W.field_message(23, body)

    @staticmethod
    def alias_payload(child: bytes, name: str) -> bytes:
        return W.field_message(1, child) + W.field_string(2, name)

    @staticmethod
    def drop(input_rel: bytes, *names: str) -> bytes:
        body = W.field_message(1, input_rel)
        for n in names:
            body += W.field_string(3, n)
        return W.field_message(21, body)

    @staticmethod
    def with_columns_renamed(input_rel: bytes, renames: dict) -> bytes:
        body = W.field_message(1, input_rel)
        for old, new in renames.items():
            body += W.field_message(3, W.field_string(1, old)
                                    + W.field_string(2, new))
        return W.field_message(19, body)

    @staticmethod
    def show_string(input_rel: bytes, num_rows: int = 20,
                    truncate: int = 20) -> bytes:
        body = (W.field_message(1, input_rel) + W.field_varint(2, num_rows)
                + W.field_varint(3, truncate))
        return W.field_message(20, body)

    @staticmethod
    def subquery_alias(input_rel: bytes, alias: str) -> bytes:
        return W.field_message(16, W.field_message(1, input_rel)
                               + W.field_string(2, alias))

    @staticmethod
    def to_df(input_rel: bytes, *names: str) -> bytes:
        body = W.field_message(1, input_rel)
        for n in names:
            body += W.field_string(2, n)
        return W.field_message(18, body)

    @staticmethod
    def tail(input_rel: bytes, n: int) -> bytes:
        return W.field_message(22, W.field_message(1, input_rel)
                               + W.field_varint(2, n))


class _Responses:
    """Raw ExecutePlan responses with ids (for reattach tests)."""

    def __init__(self, items):
        self.items = items  # [(response_id, fields)]


def _collect_batches(responses) -> "object":
    import pyarrow as pa

    batches = []
    for resp in responses:
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


def _client_relation_methods():
    def execute_relation(self, rel_bytes: bytes, operation_id=None):
        """Send a relation-tree plan; returns a pyarrow.Table."""
        plan = W.field_message(F.PLAN_ROOT, rel_bytes)
        req = (W.field_string(F.EXEC_SESSION_ID, self.session_id)
               + W.field_message(F.EXEC_PLAN, plan)
               + W.field_string(F.EXEC_OPERATION_ID,
                                operation_id or str(uuid.uuid4())))
        call = self._channel.unary_stream(
            f"/{_SERVICE}/ExecutePlan",
            request_serializer=None, response_deserializer=None)
        return _collect_batches(call(req))

    def execute_relation_raw(self, rel_bytes: bytes, operation_id: str,
                             stop_after: Optional[int] = None):
        """Like execute_relation but returns raw (response_id, msg) pairs;
        optionally abandons the stream early (reattach testing)."""
        plan = W.field_message(F.PLAN_ROOT, rel_bytes)
        req = (W.field_string(F.EXEC_SESSION_ID, self.session_id)
               + W.field_message(F.EXEC_PLAN, plan)
               + W.field_string(F.EXEC_OPERATION_ID, operation_id))
        call = self._channel.unary_stream(
            f"/{_SERVICE}/ExecutePlan",
            request_serializer=None, response_deserializer=None)
        out = []
        stream = call(req)
        for resp in stream:
            fields = W.parse(resp)
            out.append((W.first_str(fields, F.RESP_RESPONSE_ID), resp))
            if stop_after is not None and len(out) >= stop_after:
                stream.cancel()
                break
        return out

    def reattach(self, operation_id: str, last_response_id: Optional[str] = None):
        req = (W.field_string(1, self.session_id)
               + W.field_string(3, operation_id))
        if last_response_id:
            req += W.field_string(5, last_response_id)
        call = self._channel.unary_stream(
            f"/{_SERVICE}/ReattachExecute",
            request_serializer=None, response_deserializer=None)
        return [( W.first_str(W.parse(r), F.RESP_RESPONSE_ID), r)
                for r in call(req)]

    def release_until(self, operation_id: str, response_id: str):
        req = (W.field_string(1, self.session_id)
               + W.field_string(3, operation_id)
               + W.field_message(6, W.field_string(1, response_id)))
        self._call_unary("ReleaseExecute", req)

    def release_all(self, operation_id: str):
        req = (W.field_string(1, self.session_id)
               + W.field_string(3, operation_id)
               + W.field_message(5, b""))
        self._call_unary("ReleaseExecute", req)

    ConnectClient.execute_relation = execute_relation
    ConnectClient.execute_relation_raw = execute_relation_raw
    ConnectClient.reattach = reattach
    ConnectClient.release_until = release_until
    ConnectClient.release_all = release_all


_client_relation_methods()


def _client_artifact_methods():
    import zlib as _zlib

    def add_artifact(self, name: str, data: bytes):
        """Upload one artifact (single-chunk batch form)."""
        chunk = (W.field_bytes(1, data)
                 + W.field_varint(2, _zlib.crc32(data) & 0xFFFFFFFF))
        art = W.field_string(1, name) + W.field_message(2, chunk)
        batch = W.field_message(1, art)
        req = (W.field_string(1, self.session_id)
               + W.field_message(3, batch))
        call = self._channel.stream_unary(
            f"/{_SERVICE}/AddArtifacts",
            request_serializer=None, response_deserializer=None)
        resp = W.parse(call(iter([req])))
        out = []
        for s in resp.get(1, []):
            sf = W.parse(s)
            out.append((W.first_str(sf, 1), bool(W.first_varint(sf, 2))))
        return out

    def add_artifact_chunked(self, name: str, chunks):
        """Upload one artifact as a chunked stream."""
        chunks = list(chunks)
        total = sum(len(c) for c in chunks)

        def reqs():
            first = chunks[0] if chunks else b""
            ic = (W.field_bytes(1, first)
                  + W.field_varint(2, _zlib.crc32(first) & 0xFFFFFFFF))
            begin = (W.field_string(1, name) + W.field_varint(2, total)
                     + W.field_varint(3, len(chunks))
                     + W.field_message(4, ic))
            yield (W.field_string(1, self.session_id)
                   + W.field_message(4, begin))
            for c in chunks[1:]:
                ch = (W.field_bytes(1, c)
                      + W.field_varint(2, _zlib.crc32(c) & 0xFFFFFFFF))
                yield (W.field_string(1, self.session_id)
                       + W.field_message(5, ch))

        call = self._channel.stream_unary(
            f"/{_SERVICE}/AddArtifacts",
            request_serializer=None, response_deserializer=None)
        call(reqs())

    def artifact_statuses(self, names):
        req = W.field_string(1, self.session_id)
        for n in names:
            req += W.field_string(4, n)
        resp = W.parse(self._call_unary("ArtifactStatus", req))
        out = {}
        for e in resp.get(1, []):
            ef = W.parse(e)
            st = W.parse(W.first(ef, 2, b""))
            out[W.first_str(ef, 1)] = bool(W.first_varint(st, 1))
        return out

    ConnectClient.add_artifact = add_artifact
    ConnectClient.add_artifact_chunked = add_artifact_chunked
    ConnectClient.artifact_statuses = artifact_statuses


_client_artifact_methods()
