"""Spark Connect gRPC server.

Implements the SparkConnectService RPC surface over hand-rolled protobuf
wire framing (connect/wire.py) — the role of the reference's
sail-spark-connect server (ref: crates/sail-spark-connect/src/server.rs:120).

Supported today:
  * ExecutePlan — Plan.root Relation.sql and Plan.command SqlCommand /
    CreateDataFrameView; results stream back as Arrow IPC batches
  * AnalyzePlan — spark_version, schema (DDL string form), explain, tree_string
  * Config     — set / get / get_all / unset / get_option
  * Interrupt, ReleaseExecute, ReleaseSession — acknowledged no-ops
  * AddArtifacts / ArtifactStatus / ReattachExecute — unimplemented errors

Field numbers follow the public Apache Spark `spark/connect/*.proto`
definitions (Spark 3.5/4.x). The image carries no PySpark client, so
cross-client conformance is exercised by the in-repo client
(connect/client.py) which speaks the same wire subset; validating against a
live PySpark is an explicit follow-up once a client is available.
"""
from __future__ import annotations

import io
import threading
import uuid
from concurrent import futures
from typing import Dict, Iterator, Optional

import grpc

from ..engine.session import SessionContext
from . import wire as W

_SERVICE = "spark.connect.SparkConnectService"


# -- field numbers (spark/connect/base.proto) -------------------------------
class F:
    # ExecutePlanRequest
    EXEC_SESSION_ID = 1
    EXEC_PLAN = 3
    EXEC_OPERATION_ID = 6
    # Plan
    PLAN_ROOT = 1
    PLAN_COMMAND = 2
    # Relation (subset)
    REL_COMMON = 1
    REL_SQL = 10
    # SQL
    SQL_QUERY = 1
    # Command (subset)
    CMD_CREATE_VIEW = 3
    CMD_SQL = 5
    # SqlCommand
    SQLCMD_SQL = 1
    # CreateDataFrameViewCommand
    VIEW_INPUT = 1
    VIEW_NAME = 2
    VIEW_IS_GLOBAL = 3
    VIEW_REPLACE = 4
    # ExecutePlanResponse
    RESP_SESSION_ID = 1
    RESP_ARROW_BATCH = 2
    RESP_SQL_COMMAND_RESULT = 5
    RESP_OPERATION_ID = 12
    RESP_RESPONSE_ID = 13
    RESP_RESULT_COMPLETE = 14
    # ArrowBatch
    AB_ROW_COUNT = 1
    AB_DATA = 2
    # AnalyzePlanRequest oneofs
    AN_SESSION_ID = 1
    AN_SCHEMA = 4
    AN_EXPLAIN = 5
    AN_TREE_STRING = 6
    AN_IS_LOCAL = 7
    AN_IS_STREAMING = 8
    AN_INPUT_FILES = 9
    AN_SPARK_VERSION = 10
    AN_DDL_PARSE = 11
    AN_SAME_SEMANTICS = 12
    AN_SEMANTIC_HASH = 13
    AN_PERSIST = 14
    AN_UNPERSIST = 15
    AN_GET_STORAGE_LEVEL = 16
    AN_JSON_TO_DDL = 18
    # AnalyzePlanResponse oneofs
    ANR_SESSION_ID = 1
    ANR_SCHEMA = 2
    ANR_EXPLAIN = 3
    ANR_TREE_STRING = 4
    ANR_IS_LOCAL = 5
    ANR_IS_STREAMING = 6
    ANR_INPUT_FILES = 7
    ANR_SPARK_VERSION = 8
    ANR_DDL_PARSE = 9
    ANR_SAME_SEMANTICS = 10
    ANR_SEMANTIC_HASH = 11
    ANR_PERSIST = 12
    ANR_UNPERSIST = 13
    ANR_GET_STORAGE_LEVEL = 14
    ANR_JSON_TO_DDL = 16
    # ConfigRequest
    CFG_SESSION_ID = 1
    CFG_OPERATION = 3
    CFG_OP_SET = 1
    CFG_OP_GET = 2
    CFG_OP_GET_WITH_DEFAULT = 3
    CFG_OP_GET_OPTION = 4
    CFG_OP_GET_ALL = 5
    CFG_OP_UNSET = 6
    # KeyValue
    KV_KEY = 1
    KV_VALUE = 2
    # ConfigResponse
    CFGR_SESSION_ID = 1
    CFGR_PAIRS = 2


class SparkConnectServer:
    """gRPC server hosting SparkConnectService; one SessionContext per
    Spark Connect session id (ref: sail-session SessionManager)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0, device: Optional[str] = None):
        self._host = host
        self._device = device
        self._sessions: Dict[str, SessionContext] = {}
        self._session_atime: Dict[str, float] = {}
        self._lock = threading.Lock()
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=16),
            options=[("grpc.max_receive_message_length", 128 * 1024 * 1024),
                     ("grpc.max_send_message_length", 128 * 1024 * 1024)])
        handlers = {
            "ExecutePlan": grpc.unary_stream_rpc_method_handler(
                self._execute_plan, request_deserializer=None, response_serializer=None),
            "AnalyzePlan": grpc.unary_unary_rpc_method_handler(self._analyze_plan),
            "Config": grpc.unary_unary_rpc_method_handler(self._config),
            "Interrupt": grpc.unary_unary_rpc_method_handler(self._ack),
            "ReleaseExecute": grpc.unary_unary_rpc_method_handler(self._release_execute),
            "ReleaseSession": grpc.unary_unary_rpc_method_handler(self._ack),
            "ReattachExecute": grpc.unary_stream_rpc_method_handler(self._reattach_execute),
            "AddArtifacts": grpc.stream_unary_rpc_method_handler(self._add_artifacts),
            "ArtifactStatus": grpc.unary_unary_rpc_method_handler(self._artifact_status),
        }
        self._server.add_generic_rpc_handlers(
            (_GenericHandler(_SERVICE, handlers),))
        self.port = self._server.add_insecure_port(f"{host}:{port}")

    # -- lifecycle ---------------------------------------------------------
    def start(self):
        self._server.start()
        return self

    def stop(self, grace: float = 0.5):
        self._server.stop(grace)

    @property
    def address(self) -> str:
        return f"{self._host}:{self.port}"

    #: idle eviction (ref: SessionManagerActor idle timeout,
    #: crates/sail-session/src/; spark.session_timeout_secs)
    SESSION_TIMEOUT_SECS = 3600.0

    def session(self, session_id: str) -> SessionContext:
        import time as _t

        now = _t.time()
        with self._lock:
            # opportunistic eviction of idle sessions (frees their device
            # tables); ReleaseSession removes eagerly
            for sid in [k for k, ts in self._session_atime.items()
                        if now - ts > self.SESSION_TIMEOUT_SECS and k != session_id]:
                self._sessions.pop(sid, None)
                self._session_atime.pop(sid, None)
            if session_id not in self._sessions:
                self._sessions[session_id] = SessionContext(device=self._device)
            self._session_atime[session_id] = now
            return self._sessions[session_id]

    def register_session(self, session_id: str, ctx: SessionContext):
        with self._lock:
            self._sessions[session_id] = ctx

    # -- RPC impls ---------------------------------------------------------
    #: reattachable-execution buffers (ref: Executor/ExecutorBuffer,
    #: crates/sail-spark-connect/src/executor.rs:31-98): responses are
    #: retained per operation until ReleaseExecute trims them, so
    #: ReattachExecute can replay after a dropped stream.
    _BUFFER_CAP_BYTES = 64 << 20

    def _buffer_key(self, session_id: str, op_id: str):
        return (session_id, op_id)

    def _buffered_yield(self, session_id, op_id, responses):
        buf = {"responses": [], "bytes": 0, "complete": False}
        with self._lock:
            if not hasattr(self, "_op_buffers"):
                self._op_buffers = {}
            self._op_buffers[self._buffer_key(session_id, op_id)] = buf
        # drain the execution fully BEFORE streaming: a client cancelling
        # the stream must not abort production, or ReattachExecute could
        # never replay the rest (ref: the reference's Executor runs the
        # result pull in a spawned task independent of the gRPC stream)
        produced = []
        for rid, msg in responses:
            produced.append(msg)
            buf["responses"].append((rid, msg))
            buf["bytes"] += len(msg)
            while buf["bytes"] > self._BUFFER_CAP_BYTES and len(buf["responses"]) > 1:
                _, old = buf["responses"].pop(0)
                buf["bytes"] -= len(old)
        buf["complete"] = True
        for msg in produced:
            yield msg

    def _execute_plan(self, request: bytes, context) -> Iterator[bytes]:
        req = W.parse(request)
        session_id = W.first_str(req, F.EXEC_SESSION_ID)
        op_id = W.first_str(req, F.EXEC_OPERATION_ID) or str(uuid.uuid4())
        yield from self._buffered_yield(
            session_id, op_id,
            self._execute_plan_responses(req, session_id, op_id, context))

    def _execute_plan_responses(self, req, session_id, op_id, context):
        plan = W.parse(W.first(req, F.EXEC_PLAN, b""))
        sess = self.session(session_id)

        sql = None
        relation_plan = None
        is_command = False
        root = W.first(plan, F.PLAN_ROOT)
        cmd = W.first(plan, F.PLAN_COMMAND)
        if root is not None:
            rel = W.parse(root)
            sql_msg = W.first(rel, F.REL_SQL)
            if sql_msg is not None:
                sql = W.first_str(W.parse(sql_msg), F.SQL_QUERY)
            else:
                # relation-tree (DataFrame API) plan: proto -> spec ->
                # normal resolve/optimize/execute pipeline
                # (ref: crates/sail-spark-connect/src/proto/plan.rs)
                from .relations import RelationConverter
                from .relations import Unsupported as _RelUnsupported

                try:
                    spec_plan = RelationConverter(sess).convert(root)
                    relation_plan = sess.optimize(sess.resolve(spec_plan))
                except _RelUnsupported as e:
                    context.abort(grpc.StatusCode.UNIMPLEMENTED,
                                  f"relation not supported over the wire: {e}")
        elif cmd is not None:
            c = W.parse(cmd)
            sql_cmd = W.first(c, F.CMD_SQL)
            view_cmd = W.first(c, F.CMD_CREATE_VIEW)
            if sql_cmd is not None:
                sql = W.first_str(W.parse(sql_cmd), F.SQLCMD_SQL)
                is_command = True
            elif view_cmd is not None:
                v = W.parse(view_cmd)
                name = W.first_str(v, F.VIEW_NAME)
                inp = W.parse(W.first(v, F.VIEW_INPUT, b""))
                sub_sql = W.first_str(W.parse(W.first(inp, F.REL_SQL, b"")), F.SQL_QUERY)
                replace = bool(W.first(v, F.VIEW_REPLACE, 0))
                plan_ = sess.plan_sql(sub_sql)
                sess.catalog.create_view(name, plan_, replace=replace)
                yield (f"{op_id}-0", self._complete_response(session_id, op_id))
                return
            else:
                context.abort(grpc.StatusCode.UNIMPLEMENTED, "unsupported command")
        else:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "empty plan")

        if relation_plan is not None:
            from ..datasource.arrow_io import chunk_to_arrow
            from ..plan import spec as S

            if isinstance(relation_plan, S.Command):
                sess.execute_plan(relation_plan)
                table = None
            else:
                chunk = sess.execute_plan(relation_plan)
                table = chunk_to_arrow(chunk, relation_plan.schema)
        else:
            df = sess.sql(sql)
            table = df.to_arrow() if df.plan.schema else None
        rid = 0
        if table is not None:
            import pyarrow as pa

            for batch in table.to_batches(max_chunksize=65536):
                sink = io.BytesIO()
                with pa.ipc.new_stream(sink, batch.schema) as w:
                    w.write_batch(batch)
                ab = W.field_varint(F.AB_ROW_COUNT, batch.num_rows) + \
                    W.field_bytes(F.AB_DATA, sink.getvalue())
                msg = (W.field_string(F.RESP_SESSION_ID, session_id)
                       + W.field_message(F.RESP_ARROW_BATCH, ab)
                       + W.field_string(F.RESP_OPERATION_ID, op_id)
                       + W.field_string(F.RESP_RESPONSE_ID, f"{op_id}-{rid}"))
                yield (f"{op_id}-{rid}", msg)
                rid += 1
        yield (f"{op_id}-{rid}", self._complete_response(session_id, op_id, rid))

    def _complete_response(self, session_id: str, op_id: str, rid: int = 0) -> bytes:
        return (W.field_string(F.RESP_SESSION_ID, session_id)
                + W.field_string(F.RESP_OPERATION_ID, op_id)
                + W.field_string(F.RESP_RESPONSE_ID, f"{op_id}-{rid}")
                + W.field_message(F.RESP_RESULT_COMPLETE, b""))

    def _analyze_plan(self, request: bytes, context) -> bytes:
        req = W.parse(request)
        session_id = W.first_str(req, F.AN_SESSION_ID)
        sess = self.session(session_id)
        out = W.field_string(F.ANR_SESSION_ID, session_id)
        if F.AN_SPARK_VERSION in req:
            ver = W.field_string(1, "4.0.0-sail-mi355x")
            return out + W.field_message(F.ANR_SPARK_VERSION, ver)
        for fnum, rnum, render in ((F.AN_SCHEMA, F.ANR_SCHEMA, self._render_schema),
                                   (F.AN_EXPLAIN, F.ANR_EXPLAIN, self._render_explain),
                                   (F.AN_TREE_STRING, F.ANR_TREE_STRING, self._render_explain)):
            msg = W.first(req, fnum)
            if msg is not None:
                inner = W.parse(msg)
                plan = W.parse(W.first(inner, 1, b""))
                root = W.first(plan, F.PLAN_ROOT)
                sql = W.first_str(W.parse(W.first(W.parse(root or b""), F.REL_SQL, b"")),
                                  F.SQL_QUERY)
                return out + W.field_message(rnum, render(sess, sql))
        if F.AN_IS_LOCAL in req:
            return out + W.field_message(F.ANR_IS_LOCAL, W.field_varint(1, 1))
        if F.AN_IS_STREAMING in req:
            return out + W.field_message(F.ANR_IS_STREAMING, W.field_varint(1, 0))
        ddl = W.first(req, F.AN_DDL_PARSE)
        if ddl is not None:
            from ..sql.parser import parse_ddl_schema

            text = W.first_str(W.parse(ddl), 1)
            fields = parse_ddl_schema(text)
            rendered = ", ".join(f"{n} {t!r}" for n, t in fields)
            return out + W.field_message(F.ANR_DDL_PARSE, W.field_string(2, rendered))
        ss = W.first(req, F.AN_SAME_SEMANTICS)
        if ss is not None:
            inner = W.parse(ss)
            trees = []
            for fn in (1, 2):
                plan = W.parse(W.first(inner, fn, b""))
                root = W.first(plan, F.PLAN_ROOT)
                sql = W.first_str(W.parse(W.first(W.parse(root or b""), F.REL_SQL, b"")),
                                  F.SQL_QUERY)
                trees.append(self._semantic_tree(sess, sql))
            same = 1 if trees[0] == trees[1] else 0
            return out + W.field_message(F.ANR_SAME_SEMANTICS, W.field_varint(1, same))
        sh = W.first(req, F.AN_SEMANTIC_HASH)
        if sh is not None:
            inner = W.parse(sh)
            plan = W.parse(W.first(inner, 1, b""))
            root = W.first(plan, F.PLAN_ROOT)
            sql = W.first_str(W.parse(W.first(W.parse(root or b""), F.REL_SQL, b"")),
                              F.SQL_QUERY)
            import zlib

            h = zlib.crc32(self._semantic_tree(sess, sql).encode()) & 0x7FFFFFFF
            return out + W.field_message(F.ANR_SEMANTIC_HASH, W.field_varint(1, h))
        if F.AN_INPUT_FILES in req:
            inner = W.parse(W.first(req, F.AN_INPUT_FILES))
            plan = W.parse(W.first(inner, 1, b""))
            root = W.first(plan, F.PLAN_ROOT)
            sql = W.first_str(W.parse(W.first(W.parse(root or b""), F.REL_SQL, b"")),
                              F.SQL_QUERY)
            files = self._input_files(sess, sql)
            body = b"".join(W.field_string(1, f) for f in files)
            return out + W.field_message(F.ANR_INPUT_FILES, body)
        if F.AN_PERSIST in req:
            return out + W.field_message(F.ANR_PERSIST, b"")
        if F.AN_UNPERSIST in req:
            return out + W.field_message(F.ANR_UNPERSIST, b"")
        if F.AN_GET_STORAGE_LEVEL in req:
            # single-level engine: everything is device/host memory resident
            level = W.field_varint(2, 1)  # use_memory = true
            return out + W.field_message(F.ANR_GET_STORAGE_LEVEL,
                                         W.field_message(1, level))
        jtd = W.first(req, F.AN_JSON_TO_DDL)
        if jtd is not None:
            import json as _json

            text = W.first_str(W.parse(jtd), 1)
            obj = _json.loads(text)
            parts = []
            for f in obj.get("fields", []):
                parts.append(f"{f['name']} {self._json_type_ddl(f['type'])}")
            return out + W.field_message(F.ANR_JSON_TO_DDL,
                                         W.field_string(1, ",".join(parts)))
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "analyze type not supported")

    @staticmethod
    def _json_type_ddl(t) -> str:
        names = {"long": "BIGINT", "integer": "INT", "double": "DOUBLE",
                 "float": "FLOAT", "string": "STRING", "boolean": "BOOLEAN",
                 "date": "DATE", "timestamp": "TIMESTAMP", "short": "SMALLINT",
                 "byte": "TINYINT", "binary": "BINARY"}
        if isinstance(t, str):
            return names.get(t, t.upper())
        return "STRING"

    def _input_files(self, sess: SessionContext, sql: str):
        """All file paths the plan reads (ref: Spark AnalyzePlan InputFiles)."""
        from ..plan import spec as S

        plan = sess.plan_sql(sql)
        out = []

        def walk(p):
            if isinstance(p, S.DataSourceRead):
                out.extend(p.paths or [])
            for c in p.children():
                if c is not None:
                    walk(c)

        walk(plan)
        return out

    def _semantic_tree(self, sess: SessionContext, sql: str) -> str:
        from ..plan import spec as S

        return S.plan_tree_string(sess.plan_sql(sql))

    def _render_schema(self, sess: SessionContext, sql: str) -> bytes:
        plan = sess.plan_sql(sql)
        # DDL string rendering (full DataType proto encoding is a follow-up)
        ddl = ", ".join(f"{n} {t!r}" for n, t in plan.schema)
        return W.field_string(2, ddl)

    def _render_explain(self, sess: SessionContext, sql: str) -> bytes:
        from ..plan import spec as S

        plan = sess.plan_sql(sql)
        return W.field_string(1, S.plan_tree_string(plan))

    def _config(self, request: bytes, context) -> bytes:
        req = W.parse(request)
        session_id = W.first_str(req, F.CFG_SESSION_ID)
        sess = self.session(session_id)
        op = W.parse(W.first(req, F.CFG_OPERATION, b""))
        pairs_out = b""
        if W.first(op, F.CFG_OP_SET) is not None:
            st = W.parse(W.first(op, F.CFG_OP_SET))
            for kv in st.get(1, []):
                kvf = W.parse(kv)
                sess.conf[W.first_str(kvf, F.KV_KEY)] = W.first_str(kvf, F.KV_VALUE)
        elif W.first(op, F.CFG_OP_GET) is not None or W.first(op, F.CFG_OP_GET_OPTION) is not None:
            g = W.parse(W.first(op, F.CFG_OP_GET) or W.first(op, F.CFG_OP_GET_OPTION))
            for key in g.get(1, []):
                k = key.decode()
                v = sess.conf.get(k)
                kv = W.field_string(F.KV_KEY, k)
                if v is not None:
                    kv += W.field_string(F.KV_VALUE, v)
                pairs_out += W.field_message(F.CFGR_PAIRS, kv)
        elif W.first(op, F.CFG_OP_GET_ALL) is not None:
            for k, v in sess.conf.items():
                kv = W.field_string(F.KV_KEY, k) + W.field_string(F.KV_VALUE, v)
                pairs_out += W.field_message(F.CFGR_PAIRS, kv)
        elif W.first(op, F.CFG_OP_UNSET) is not None:
            u = W.parse(W.first(op, F.CFG_OP_UNSET))
            for key in u.get(1, []):
                sess.conf.pop(key.decode(), None)
        return W.field_string(F.CFGR_SESSION_ID, session_id) + pairs_out

    def _reattach_execute(self, request: bytes, context) -> Iterator[bytes]:
        """Replay buffered responses after last_response_id (ref: the
        reference's reattachable-execution state machine)."""
        req = W.parse(request)
        session_id = W.first_str(req, 1)
        op_id = W.first_str(req, 3)
        last = W.first_str(req, 5)
        buf = getattr(self, "_op_buffers", {}).get(
            self._buffer_key(session_id, op_id))
        if buf is None:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "INVALID_HANDLE.OPERATION_NOT_FOUND")
        start = 0
        if last:
            for i, (rid, _) in enumerate(buf["responses"]):
                if rid == last:
                    start = i + 1
                    break
        for _, msg in list(buf["responses"][start:]):
            yield msg

    def _release_execute(self, request: bytes, context) -> bytes:
        req = W.parse(request)
        session_id = W.first_str(req, 1)
        op_id = W.first_str(req, 3)
        key = self._buffer_key(session_id, op_id)
        buf = getattr(self, "_op_buffers", {}).get(key)
        if buf is not None:
            until = W.first(req, 6)
            if until is not None:
                rid = W.first_str(W.parse(until), 1)
                keep = buf["responses"]
                for i, (r, _) in enumerate(keep):
                    if r == rid:
                        dropped = keep[:i + 1]
                        buf["responses"] = keep[i + 1:]
                        buf["bytes"] -= sum(len(m) for _, m in dropped)
                        break
            if W.first(req, 5) is not None:  # release_all
                self._op_buffers.pop(key, None)
        return (W.field_string(1, session_id)
                + W.field_string(2, op_id))

    def _session_artifacts(self, session_id: str) -> dict:
        with self._lock:
            if not hasattr(self, "_artifacts"):
                self._artifacts = {}
            return self._artifacts.setdefault(session_id, {})

    def _add_artifacts(self, request_iterator, context) -> bytes:
        """Client-streamed artifact upload (single-chunk batches and
        chunked artifacts; ref: sail-spark-connect server.rs AddArtifacts).
        Artifacts are retained per session (pyfile/jar payloads are stored;
        execution-side artifact use is the Python UDF registry)."""
        import zlib as _zlib

        session_id = ""
        summaries = []
        pending_name = None
        pending_buf = b""
        store = None
        for raw in request_iterator:
            req = W.parse(raw)
            session_id = W.first_str(req, 1) or session_id
            if store is None:
                store = self._session_artifacts(session_id)
            batch = W.first(req, 3)
            if batch is not None:
                for art in W.parse(batch).get(1, []):
                    af = W.parse(art)
                    name = W.first_str(af, 1)
                    ch = W.parse(W.first(af, 2, b""))
                    data = W.first(ch, 1, b"")
                    crc = W.first_varint(ch, 2, 0)
                    ok = (_zlib.crc32(data) & 0xFFFFFFFF) == crc or crc == 0
                    if ok:
                        store[name] = data
                    summaries.append((name, ok))
            begin = W.first(req, 4)
            if begin is not None:
                bf = W.parse(begin)
                pending_name = W.first_str(bf, 1)
                pending_buf = b""
                ic = W.first(bf, 4)  # BeginChunkedArtifact.initial_chunk
                if ic is not None:
                    pending_buf += W.first(W.parse(ic), 1, b"")
            chunk = W.first(req, 5)
            if chunk is not None and pending_name is not None:
                pending_buf += W.first(W.parse(chunk), 1, b"")
        if pending_name is not None:
            store = store if store is not None else \
                self._session_artifacts(session_id)
            store[pending_name] = pending_buf
            summaries.append((pending_name, True))
        out = W.field_string(2, session_id)
        for name, ok in summaries:
            body = W.field_string(1, name) + W.field_varint(2, 1 if ok else 0)
            out += W.field_message(1, body)
        return out

    def _artifact_status(self, request: bytes, context) -> bytes:
        req = W.parse(request)
        session_id = W.first_str(req, 1)
        store = self._session_artifacts(session_id)
        out = W.field_string(2, session_id)
        for name in req.get(4, []):
            n = name.decode()
            status = W.field_varint(1, 1 if n in store else 0)
            entry = (W.field_string(1, n) + W.field_message(2, status))
            out += W.field_message(1, entry)
        return out

    def _ack(self, request: bytes, context) -> bytes:
        return b""

    def _unimplemented_unary(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "not implemented yet")

    def _unimplemented_stream(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "not implemented yet")
        yield b""  # pragma: no cover


class _GenericHandler(grpc.GenericRpcHandler):
    def __init__(self, service: str, handlers):
        self._service = service
        self._handlers = handlers

    def service(self, handler_call_details):
        name = handler_call_details.method.rsplit("/", 1)[-1]
        svc = handler_call_details.method.rsplit("/", 2)[-2] if "/" in handler_call_details.method else ""
        if svc != self._service:
            return None
        return self._handlers.get(name)
