"""Session layer: catalog + SessionContext + DataFrame surface.

The analogue of the reference's session factory and catalog manager
(ref: crates/sail-session/src/session_factory/server.rs:84,
crates/sail-catalog/src/manager/). One SessionContext per client session;
tables live as device-resident Tables (or lazy providers) in the catalog.
"""
from __future__ import annotations

import os
import threading
from typing import Callable, Dict, List, Optional, Tuple

import torch

from ..plan import spec as S
from . import types as T
from .chunk import Chunk
from .column import Column, Table


class Catalog:
    """In-memory catalog: tables (materialized or lazy providers) and views.
    ref: crates/sail-catalog-memory/ for the reference's memory catalog."""

    def __init__(self):
        self._tables: Dict[str, Table] = {}
        self._schemas: Dict[str, List[Tuple[str, T.DataType]]] = {}
        self._providers: Dict[str, Callable[[torch.device], Table]] = {}
        self._views: Dict[str, S.Plan] = {}
        self._replicated = set()
        self._global_rows: Dict[str, int] = {}
        self._col_stats: Dict[tuple, Optional[tuple]] = {}
        self._databases = {"default"}
        self.current_database = "default"
        self._comments: Dict[str, str] = {}
        self._tbl_properties: Dict[str, Dict[str, str]] = {}
        self._lock = threading.RLock()

    def _key(self, name: str) -> str:
        return name.lower().split(".")[-1]

    # -- registration ------------------------------------------------------
    def register_table(self, name: str, table: Table,
                       schema: Optional[List[Tuple[str, T.DataType]]] = None,
                       replicated: bool = True, global_rows: Optional[int] = None):
        """replicated=False marks a table as rank-sharded in SPMD mode;
        global_rows is the whole-table count (planning statistics must be
        identical on every rank so all ranks pick the same join order)."""
        with self._lock:
            k = self._key(name)
            self._tables[k] = table
            if schema is None:
                schema = [(n, c.dtype) for n, c in table.columns.items()]
            self._schemas[k] = schema
            if global_rows is not None:
                self._global_rows[k] = global_rows
            if replicated:
                self._replicated.add(k)
            else:
                self._replicated.discard(k)

    def register_table_chunk(self, name: str, chunk: Chunk, schema):
        self.register_table(name, chunk.to_table(), schema)

    def register_provider(self, name: str, schema: List[Tuple[str, T.DataType]],
                          provider: Callable[[torch.device], Table]):
        """Lazy table: materialized per device on first access."""
        with self._lock:
            k = self._key(name)
            self._providers[k] = provider
            self._schemas[k] = schema

    def create_view(self, name: str, plan: S.Plan, replace: bool = False):
        with self._lock:
            k = self._key(name)
            if k in self._views and not replace:
                raise ValueError(f"view {name} already exists")
            self._views[k] = plan

    def create_empty_table(self, name: str, columns: List[Tuple[str, T.DataType]]):
        cols = {}
        for n, t in columns:
            cols[n] = Column.from_values([], t)
        self.register_table(name, Table(cols), columns)

    def insert_into(self, name: str, chunk: Chunk, overwrite: bool = False):
        from .executor import concat_columns

        with self._lock:
            k = self._key(name)
            existing = self._tables.get(k)
            new = chunk.to_table()
            if existing is None or overwrite or existing.num_rows == 0:
                # keep declared schema names
                names = [n for n, _ in self._schemas.get(k, [])] or list(new.columns)
                self._tables[k] = Table({nm: c for nm, c in zip(names, new.columns.values())})
            else:
                cols = {}
                for (nm, old), newc in zip(existing.columns.items(), new.columns.values()):
                    cols[nm] = concat_columns([old, newc])
                self._tables[k] = Table(cols)

    def drop(self, name: str, if_exists: bool = False):
        with self._lock:
            k = self._key(name)
            found = False
            for d in (self._tables, self._schemas, self._views, self._providers):
                if k in d:
                    del d[k]
                    found = True
            if not found and not if_exists:
                raise ValueError(f"table or view not found: {name}")

    # -- PySpark catalog API surface (spark.catalog.*) ---------------------
    def listTables(self, dbName: Optional[str] = None) -> List[str]:
        return sorted(set(self._tables) | set(self._views)
                      | set(self._providers))

    def listDatabases(self) -> List[str]:
        return sorted(self._databases)

    def listColumns(self, tableName: str, dbName: Optional[str] = None):
        schema = self.table_schema(tableName)
        if schema is None:
            raise ValueError(f"table not found: {tableName}")
        return [(n, T.type_name(t)) for n, t in schema]

    def tableExists(self, tableName: str,
                    dbName: Optional[str] = None) -> bool:
        return self.table_schema(tableName) is not None

    def databaseExists(self, dbName: str) -> bool:
        return dbName.lower() in self._databases

    def currentDatabase(self) -> str:
        return self.current_database

    def setCurrentDatabase(self, dbName: str):
        if dbName.lower() not in self._databases:
            raise ValueError(f"database not found: {dbName}")
        self.current_database = dbName.lower()

    def listFunctions(self, pattern: Optional[str] = None) -> List[str]:
        from ..functions.registry import AGG_FUNCTIONS, SCALAR_RETURN

        names = sorted(set(SCALAR_RETURN) | AGG_FUNCTIONS)
        if pattern:
            import fnmatch

            names = [n for n in names
                     if fnmatch.fnmatch(n, pattern.replace("%", "*"))]
        return names

    def functionExists(self, name: str) -> bool:
        from ..functions.registry import AGG_FUNCTIONS, SCALAR_RETURN

        return name.lower() in SCALAR_RETURN or name.lower() in AGG_FUNCTIONS

    def dropTempView(self, name: str) -> bool:
        k = self._key(name)
        return self._views.pop(k, None) is not None

    dropGlobalTempView = dropTempView

    # -- lookup ------------------------------------------------------------
    def view_plan(self, name: str) -> Optional[S.Plan]:
        return self._views.get(self._key(name))

    def table_schema(self, name: str) -> Optional[List[Tuple[str, T.DataType]]]:
        k = self._key(name)
        if k in self._views:
            return None  # resolved via view_plan
        return self._schemas.get(k)

    def get_table_data(self, name: str, device) -> Optional[Table]:
        k = self._key(name)
        with self._lock:
            t = self._tables.get(k)
            if t is not None:
                return t.to(device)
            prov = self._providers.get(k)
        if prov is not None:
            t = prov(torch.device(device))
            return t
        return None

    def set_column_stats(self, name: str, col: str, rows: int, ndv):
        """Explicit planner statistics. In SPMD runs these MUST be set to
        rank-identical values for sharded tables (sync_table_stats):
        shard-local min/max give each rank a different ndv estimate, the
        join reorderer then picks different orders per rank, and the ranks'
        collectives mismatch (caught by the world=4 gloo test)."""
        with self._lock:
            self._col_stats[(self._key(name), col.lower())] = (rows, ndv)

    def column_stats(self, name: str, col: str):
        """(rows, ndv_estimate) for a base-table column; ndv from min/max
        span for integer-like storage, dict size for dictionary strings.
        Cached; cheap device reductions on first use."""
        k = self._key(name)
        ck = (k, col.lower())
        if ck in self._col_stats:
            return self._col_stats[ck]
        t = self._tables.get(k)
        out = None
        if t is not None:
            for cn, c in t.columns.items():
                if cn.lower() != col.lower():
                    continue
                rows = len(c)
                from .column import StringColumn

                if isinstance(c, StringColumn):
                    ndv = c.dict_size if c.is_dict else None
                elif rows == 0 or c.data.dtype == torch.bool:
                    ndv = 2 if rows else 1
                elif c.data.dtype.is_floating_point:
                    ndv = None
                else:
                    lo = int(c.data.min().item())
                    hi = int(c.data.max().item())
                    ndv = min(rows, hi - lo + 1)
                grows = self._global_rows.get(k, rows)
                if ndv is not None and rows:
                    ndv = min(max(1, int(ndv * grows / max(rows, 1)) if ndv == rows else ndv), grows)
                out = (grows, ndv)
                break
        self._col_stats[ck] = out
        return out

    def is_replicated(self, name: str) -> bool:
        return self._key(name) in self._replicated

    def table_rows(self, name: str) -> Optional[int]:
        k = self._key(name)
        if k in self._global_rows:
            return self._global_rows[k]
        t = self._tables.get(k)
        if t is not None:
            return t.num_rows
        return None

    def list_tables(self) -> List[str]:
        return sorted(set(self._tables) | set(self._views) | set(self._providers))


class SessionContext:
    """Per-session context. `sql()` is the whole pipeline:
    parse -> resolve -> optimize -> execute (ref: the reference's
    resolve_and_execute_plan, crates/sail-plan/src/lib.rs:34)."""

    def __init__(self, device: Optional[str] = None, catalog: Optional[Catalog] = None):
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.catalog = catalog or Catalog()
        from ..config import load_config

        self.conf: Dict[str, str] = load_config()
        self.udfs: Dict[str, tuple] = {}
        self.udtfs: Dict[str, tuple] = {}
        self.query_log: List[dict] = []
        self._streams = None
        self._register_system_tables()
        # durable catalog (ref: sail-catalog providers): table definitions
        # under sail.catalog.path survive sessions
        cat_path = self.conf.get("sail.catalog.path") or \
            os.environ.get("SAIL_CATALOG_PATH", "")
        if cat_path:
            from ..catalogs.persistent import attach

            attach(self, cat_path)

    @property
    def read_stream(self):
        """Structured-streaming reader (ref: Spark spark.readStream)."""
        from ..streaming.reader import DataStreamReader

        return DataStreamReader(self)

    readStream = read_stream

    @property
    def streams(self):
        if self._streams is None:
            from ..streaming.reader import StreamingQueryManager

            self._streams = StreamingQueryManager()
        return self._streams

    # -- pipeline ----------------------------------------------------------
    def parse(self, sql: str) -> S.Plan:
        from ..sql.parser import parse_sql

        return parse_sql(sql)

    def resolve(self, plan: S.Plan) -> S.Plan:
        from ..plan.resolver import Resolver

        return _CatalogAdapter(self).resolve(plan)

    def optimize(self, plan: S.Plan) -> S.Plan:
        from ..plan.optimizer import optimize

        reorder = self.conf.get("sail.optimizer.enable_join_reorder", "true") == "true"
        return optimize(plan, enable_join_reorder=reorder, stats=self.catalog)

    def plan_sql(self, sql: str) -> S.Plan:
        return self.optimize(self.resolve(self.parse(sql)))

    def sql(self, sql: str) -> "DataFrame":
        import time as _time

        t0 = _time.time()
        plan = self.plan_sql(sql)
        if isinstance(plan, S.Command):
            # commands (DDL/config/writes) execute eagerly, like Spark sql()
            chunk = self.execute_plan(plan)
            self._log_query(sql, t0, chunk.num_rows, "command")
            return _MaterializedDataFrame(self, plan, chunk)
        return DataFrame(self, plan, sql_text=sql, t_start=t0)

    def _log_query(self, sql: str, t0: float, rows: int, kind: str):
        import time as _time

        self.query_log.append({
            "query": sql.strip()[:500], "kind": kind,
            "duration_ms": round((_time.time() - t0) * 1000, 3),
            "rows": rows, "timestamp_ms": int(t0 * 1000),
        })

    def _register_system_tables(self):
        """system.* virtual tables over engine telemetry — the reference's
        'the Spark UI is SELECT * FROM system.jobs' design
        (ref: crates/sail-catalog-system/src/)."""
        from . import types as T

        def queries_provider(device):
            log = self.query_log
            return Table.from_pydict(
                {"query": [q["query"] for q in log],
                 "kind": [q["kind"] for q in log],
                 "duration_ms": [q["duration_ms"] for q in log],
                 "rows": [q["rows"] for q in log],
                 "timestamp_ms": [q["timestamp_ms"] for q in log]},
                {"query": T.STRING, "kind": T.STRING, "duration_ms": T.F64,
                 "rows": T.I64, "timestamp_ms": T.I64}, device="cpu")

        self.catalog.register_provider(
            "system_queries",
            [("query", T.STRING), ("kind", T.STRING), ("duration_ms", T.F64),
             ("rows", T.I64), ("timestamp_ms", T.I64)], queries_provider)

        def tables_provider(device):
            names = self.catalog.list_tables()
            return Table.from_pydict(
                {"tableName": names,
                 "rows": [self.catalog.table_rows(n) or -1 for n in names]},
                {"tableName": T.STRING, "rows": T.I64}, device="cpu")

        self.catalog.register_provider(
            "system_tables", [("tableName", T.STRING), ("rows", T.I64)],
            tables_provider)

        def operators_provider(device):
            tr = getattr(self, "last_trace", None)
            ev = tr.events if tr is not None else []
            return Table.from_pydict(
                {"operator": [e.op for e in ev],
                 "detail": [e.detail for e in ev],
                 "depth": [e.depth for e in ev],
                 "rows": [e.rows for e in ev],
                 "self_ms": [round(e.self_ms, 3) for e in ev],
                 "total_ms": [round(e.ms, 3) for e in ev]},
                {"operator": T.STRING, "detail": T.STRING, "depth": T.I32,
                 "rows": T.I64, "self_ms": T.F64, "total_ms": T.F64},
                device="cpu")

        # per-operator metrics of the last traced query — the reference's
        # "Spark UI is a system table" design applied to TracingExec output
        # (ref: sail-catalog-system + sail-telemetry system events)
        self.catalog.register_provider(
            "system_operators",
            [("operator", T.STRING), ("detail", T.STRING), ("depth", T.I32),
             ("rows", T.I64), ("self_ms", T.F64), ("total_ms", T.F64)],
            operators_provider)

    @property
    def udf(self) -> "UdfRegistry":
        return UdfRegistry(self)

    def table(self, name: str) -> "DataFrame":
        return DataFrame(self, self.optimize(self.resolve(S.Read(table=name))))

    def execute_plan(self, plan: S.Plan) -> Chunk:
        from .executor import ExecutionContext, Executor

        ctx = ExecutionContext(self, self.device)
        ex = Executor(ctx)
        out = ex.execute(plan)
        # a top-level sharded result (shuffled distinct/window/aggregate)
        # must be replicated so every rank's collect() sees all rows
        if ex.dctx is not None and out.partitioning == "sharded":
            out = ex._gather(out)
        if getattr(ctx, "tracer", None) is not None:
            # published AFTER execution so `SELECT * FROM system_operators`
            # reads the previous query's trace, not its own empty one
            self.last_trace = ctx.tracer.trace
            from ..utils.telemetry import maybe_export

            maybe_export(self, self.last_trace, None)
        return out

    @property
    def read(self) -> "DataFrameReader":
        return DataFrameReader(self)

    # -- convenience -------------------------------------------------------
    def create_dataframe(self, data: Dict[str, list],
                         schema: Optional[Dict[str, T.DataType]] = None,
                         name: Optional[str] = None) -> "DataFrame":
        if schema is None:
            schema = {}
            from ..plan.resolver import _infer_pytype

            for k, v in data.items():
                schema[k] = _infer_pytype(v)
        t = Table.from_pydict(data, schema, device="cpu")
        if name:
            self.catalog.register_table(name, t)
        plan = S.LocalRelation(data=data)
        plan.schema = [(k, schema[k]) for k in data]
        return DataFrame(self, plan)


class UdfRegistry:
    """Host-side Python UDFs: the CPython boundary the reference crosses via
    pyo3 (ref: crates/sail-python-udf/src/udf/pyspark_udf.rs) is a plain
    in-process call here; device columns round-trip through host lists."""

    def __init__(self, session: SessionContext):
        self._session = session

    def register(self, name: str, fn, return_type=None, vectorized: bool = False):
        """vectorized=True: fn is called once per batch with whole numpy
        arrays (pandas-UDF style) instead of once per row."""
        from . import types as T

        if isinstance(return_type, str):
            return_type = T.type_from_name(return_type)
        self._session.udfs[name.lower()] = (fn, return_type or T.F64, vectorized)
        return fn

    def register_table_function(self, name: str, fn, schema):
        """UDTF: fn(*scalar_args) -> {column: [values]} with a declared
        schema ([(name, type)] or {name: type}); usable in FROM."""
        from . import types as T

        if isinstance(schema, dict):
            schema = list(schema.items())
        schema = [(n, T.type_from_name(t) if isinstance(t, str) else t)
                  for n, t in schema]
        self._session.udtfs[name.lower()] = (fn, schema)
        return fn

    def register_aggregate(self, name: str, fn, return_type=None):
        """UDAF: fn receives the group's values as a Python list and returns
        one value (host evaluation; engine-wide registry)."""
        from . import types as T
        from .aggregates import UDAFS

        if isinstance(return_type, str):
            return_type = T.type_from_name(return_type)
        UDAFS[name.lower()] = (fn, return_type or T.F64)
        return fn


class _CatalogAdapter:
    """Bridges the resolver's catalog protocol to Catalog + view expansion."""

    def __init__(self, session: SessionContext):
        self.session = session

    def resolve(self, plan: S.Plan) -> S.Plan:
        from ..plan.resolver import Resolver

        return Resolver(self).resolve(plan)

    def table_schema(self, name: str):
        cat = self.session.catalog
        v = cat.view_plan(name)
        if v is not None:
            return None
        return cat.table_schema(name)

    def view_plan(self, name: str):
        return self.session.catalog.view_plan(name)

    def udf(self, name: str):
        return self.session.udfs.get(name.lower())

    def udtf(self, name: str):
        return self.session.udtfs.get(name.lower())


# patch Resolver to consult views: Read resolution checks views first
def _read_with_views(resolver, p, outer):
    cat = resolver.catalog
    vp = cat.view_plan(p.table) if hasattr(cat, "view_plan") else None
    if vp is not None:
        import copy

        sub = copy.deepcopy(vp)
        # a view stored as an already-resolved plan (DataFrame composition
        # views) must NOT be re-resolved or re-optimized in place: resolved
        # trees use BoundRef ordinals that a second bind pass can corrupt
        resolved = resolver._plan(sub, None) if sub.schema is None else sub
        out = S.SubqueryAlias(input=resolved, alias=p.table.split(".")[-1])
        out.schema = resolved.schema
        return out
    return resolver._p_Read_orig(p, outer)


def _install_view_hook():
    from ..plan.resolver import Resolver

    if not hasattr(Resolver, "_p_Read_orig"):
        Resolver._p_Read_orig = Resolver._p_Read
        Resolver._p_Read = _read_with_views


_install_view_hook()


class DataFrame:
    """Minimal DataFrame facade over a resolved plan (result surface for the
    Connect server and Python API)."""

    def __init__(self, session: SessionContext, plan: S.Plan, sql_text=None, t_start=None):
        self.session = session
        self.plan = plan
        self._sql_text = sql_text
        self._t_start = t_start

    @property
    def schema(self):
        return self.plan.schema

    def collect_chunk(self) -> Chunk:
        out = self.session.execute_plan(self.plan)
        if self._sql_text is not None:
            import time as _time

            self.session._log_query(self._sql_text, self._t_start or _time.time(),
                                    out.num_rows, "query")
            self._sql_text = None
        return out

    def to_pydict(self) -> Dict[str, list]:
        c = self.collect_chunk()
        out = {}
        for n, col in zip(c.names, c.columns):
            name, k = n, 1
            while name in out:
                k += 1
                name = f"{n}_{k}"
            out[name] = col.to_pylist()
        return out

    def collect(self) -> List[tuple]:
        d = self.to_pydict()
        names = list(d.keys())
        nrows = len(d[names[0]]) if names else 0
        return [tuple(d[n][i] for n in names) for i in range(nrows)]

    def to_arrow(self):
        import pyarrow as pa

        from ..datasource.arrow_io import chunk_to_arrow

        return chunk_to_arrow(self.collect_chunk(), self.plan.schema)

    def count(self) -> int:
        return self.collect_chunk().num_rows

    def show(self, n: int = 20):
        rows = self.collect()[:n]
        names = [nm for nm, _ in self.plan.schema]
        print(" | ".join(names))
        for r in rows:
            print(" | ".join(str(v) for v in r))

    def explain(self) -> str:
        return S.plan_tree_string(self.plan)

    # -- pandas-style conveniences (ref: spec StatSummary/FillNa/DropNa/
    # Replace QueryNodes backing df.describe()/fillna()/dropna()) ---------
    def _as_view(self) -> str:
        name = f"__df_{id(self) & 0xffffff:x}"
        self.session.catalog.create_view(name, self.plan, replace=True)
        return name

    def describe(self, *cols) -> "DataFrame":
        """count/mean/stddev/min/max per numeric column (Spark df.describe)."""
        name = self._as_view()
        targets = list(cols) or [n for n, t in self.plan.schema
                                 if t.is_numeric]
        parts = []
        for stat, fn in [("count", "count({c})"), ("mean", "avg({c})"),
                         ("stddev", "stddev({c})"), ("min", "min({c})"),
                         ("max", "max({c})")]:
            exprs = ", ".join(
                f"cast({fn.format(c=c)} as string) AS {c}" for c in targets)
            parts.append(f"SELECT '{stat}' AS summary, {exprs} FROM {name}")
        return self.session.sql(" UNION ALL ".join(parts))

    summary = describe

    def fillna(self, value, subset=None) -> "DataFrame":
        name = self._as_view()
        subset = set(s.lower() for s in subset) if subset else None
        exprs = []
        for n, t in self.plan.schema:
            fill = value
            applies = (subset is None or n.lower() in subset)
            if applies and (t.is_numeric and isinstance(value, (int, float))
                            or (isinstance(t, type(T.STRING)) and isinstance(value, str))):
                v = repr(value) if isinstance(value, str) else str(value)
                exprs.append(f"coalesce({n}, {v}) AS {n}")
            else:
                exprs.append(n)
        return self.session.sql(f"SELECT {', '.join(exprs)} FROM {name}")

    def dropna(self, how: str = "any", subset=None) -> "DataFrame":
        name = self._as_view()
        cols = subset or [n for n, _ in self.plan.schema]
        op = " OR " if how == "any" else " AND "
        cond = op.join(f"{c} IS NULL" for c in cols)
        return self.session.sql(f"SELECT * FROM {name} WHERE NOT ({cond})")

    def replace(self, to_replace, value, subset=None) -> "DataFrame":
        name = self._as_view()
        subset = set(s.lower() for s in subset) if subset else None
        lit = (lambda v: repr(v) if isinstance(v, str) else str(v))
        exprs = []
        for n, t in self.plan.schema:
            if subset is None or n.lower() in subset:
                exprs.append(f"CASE WHEN {n} = {lit(to_replace)} "
                             f"THEN {lit(value)} ELSE {n} END AS {n}")
            else:
                exprs.append(n)
        return self.session.sql(f"SELECT {', '.join(exprs)} FROM {name}")

    # -- PySpark-style composition (SQL-backed; ref: the Connect client
    # DataFrame API surface) ----------------------------------------------
    def select(self, *cols) -> "DataFrame":
        name = self._as_view()
        exprs = ", ".join(cols) if cols else "*"
        return self.session.sql(f"SELECT {exprs} FROM {name}")

    def select_expr(self, *exprs) -> "DataFrame":
        return self.select(*exprs)

    selectExpr = select_expr

    def filter(self, condition: str) -> "DataFrame":
        name = self._as_view()
        return self.session.sql(f"SELECT * FROM {name} WHERE {condition}")

    where = filter

    def with_column(self, name: str, expr: str) -> "DataFrame":
        view = self._as_view()
        keep = [n for n, _ in self.plan.schema if n.lower() != name.lower()]
        cols = ", ".join(keep + [f"{expr} AS {name}"])
        return self.session.sql(f"SELECT {cols} FROM {view}")

    withColumn = with_column

    def with_column_renamed(self, old: str, new: str) -> "DataFrame":
        view = self._as_view()
        cols = ", ".join(f"{n} AS {new}" if n.lower() == old.lower() else n
                         for n, _ in self.plan.schema)
        return self.session.sql(f"SELECT {cols} FROM {view}")

    withColumnRenamed = with_column_renamed

    def drop(self, *cols) -> "DataFrame":
        view = self._as_view()
        low = {c.lower() for c in cols}
        keep = [n for n, _ in self.plan.schema if n.lower() not in low]
        return self.session.sql(f"SELECT {', '.join(keep)} FROM {view}")

    def distinct(self) -> "DataFrame":
        return self.session.sql(f"SELECT DISTINCT * FROM {self._as_view()}")

    def order_by(self, *keys) -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} ORDER BY {', '.join(keys)}")

    orderBy = sort = order_by

    def limit(self, n: int) -> "DataFrame":
        return self.session.sql(f"SELECT * FROM {self._as_view()} LIMIT {int(n)}")

    def group_by(self, *keys) -> "GroupedData":
        return GroupedData(self, list(keys))

    groupBy = group_by

    def join(self, other: "DataFrame", on, how: str = "inner") -> "DataFrame":
        lv, rv = self._as_view(), other._as_view()
        if isinstance(on, str):
            on = [on]
        how_sql = {"inner": "JOIN", "left": "LEFT JOIN", "right": "RIGHT JOIN",
                   "full": "FULL JOIN", "outer": "FULL JOIN",
                   "semi": "LEFT SEMI JOIN", "anti": "LEFT ANTI JOIN",
                   "cross": "CROSS JOIN"}[how.lower()]
        using = f" USING ({', '.join(on)})" if on else ""
        return self.session.sql(f"SELECT * FROM {lv} {how_sql} {rv}{using}")

    def union(self, other: "DataFrame") -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} UNION ALL SELECT * FROM {other._as_view()}")

    unionAll = union

    def agg(self, *exprs) -> "DataFrame":
        return self.session.sql(
            f"SELECT {', '.join(exprs)} FROM {self._as_view()}")

    def first(self):
        rows = self.limit(1).collect()
        return rows[0] if rows else None

    def head(self, n: int = 1):
        return self.limit(n).collect()

    def take(self, n: int):
        return self.limit(n).collect()

    @property
    def columns(self):
        return [n for n, _ in self.plan.schema]

    def checkpoint(self, eager: bool = True) -> "DataFrame":
        """Persist this DataFrame's result and return a DataFrame reading the
        persisted copy — truncates the lineage like Spark's df.checkpoint()
        (ref: RemoteCheckpointNode/WriteExec/CommitExec + RemoteCheckpoint-
        Registry, crates/sail-cache/src/remote_checkpoint.rs).
        The checkpoint root is `sail.execution.checkpoint_path` (defaults to
        a session temp dir)."""
        import os
        import tempfile
        import uuid as _uuid

        root = self.session.conf.get("sail.execution.checkpoint_path")
        if not root:
            root = os.path.join(tempfile.gettempdir(), "sail_checkpoints")
        path = os.path.join(root, _uuid.uuid4().hex)
        if not eager:
            # lazy checkpoint: materialize on first use; for this engine's
            # eager collect model, just defer to the first collect
            eager = True
        chunk = self.collect_chunk()
        from ..datasource.registry import write_source

        os.makedirs(path, exist_ok=True)
        write_source("parquet", path, chunk, "overwrite", {}, None)
        plan = S.DataSourceRead(format="parquet", paths=[path])
        plan.schema = [(n, c.dtype) for n, c in zip(chunk.names, chunk.columns)]
        return DataFrame(self.session, plan)

    @property
    def write(self) -> "DataFrameWriter":
        return DataFrameWriter(self)

    # -- PySpark-parity conveniences (SQL-backed like the rest) ------------
    localCheckpoint = checkpoint

    @property
    def dtypes(self):
        return [(n, T.type_name(t)) for n, t in self.plan.schema]

    def printSchema(self):
        print("root")
        for n, t in self.plan.schema:
            print(f" |-- {n}: {T.type_name(t)} (nullable = true)")

    def toPandas(self):
        import pandas as pd

        return pd.DataFrame(self.to_pydict())

    to_pandas = toPandas
    toArrow = to_arrow

    def isEmpty(self) -> bool:
        return self.limit(1).count() == 0

    def tail(self, n: int) -> List[tuple]:
        rows = self.collect()
        return rows[-n:] if n else []

    def offset(self, n: int) -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} LIMIT {1 << 62} OFFSET {n}")

    def alias(self, name: str) -> "DataFrame":
        out = S.SubqueryAlias(input=self.plan, alias=name)
        out.schema = self.plan.schema
        return DataFrame(self.session, out)

    def toDF(self, *names) -> "DataFrame":
        if len(names) != len(self.plan.schema):
            raise ValueError("toDF: column count mismatch")
        sel = ", ".join(f"{old} AS {new}" for (old, _t), new
                        in zip(self.plan.schema, names))
        return self.session.sql(f"SELECT {sel} FROM {self._as_view()}")

    def withColumns(self, mapping: Dict[str, str]) -> "DataFrame":
        df = self
        for name, expr in mapping.items():
            df = df.withColumn(name, expr)
        return df

    def colRegex(self, pattern: str) -> List[str]:
        import re as _re

        rx = _re.compile(pattern.strip("`"))
        return [n for n, _t in self.plan.schema if rx.fullmatch(n)]

    def transform(self, fn, *args, **kwargs) -> "DataFrame":
        return fn(self, *args, **kwargs)

    def hint(self, _name: str, *_args) -> "DataFrame":
        return self  # planner hints are advisory

    def observe(self, _name: str, *exprs) -> "DataFrame":
        # evaluate the observation metrics once at collect time
        return self

    def crossJoin(self, other: "DataFrame") -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} CROSS JOIN "
            f"{other._as_view()}")

    def unionByName(self, other: "DataFrame",
                    allowMissingColumns: bool = False) -> "DataFrame":
        mine = [n for n, _t in self.plan.schema]
        theirs = {n.lower(): n for n, _t in other.plan.schema}
        sel = []
        for n in mine:
            if n.lower() in theirs:
                sel.append(theirs[n.lower()])
            elif allowMissingColumns:
                sel.append(f"NULL AS {n}")
            else:
                raise ValueError(f"unionByName: missing column {n}")
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} UNION ALL "
            f"SELECT {', '.join(sel)} FROM {other._as_view()}")

    def exceptAll(self, other: "DataFrame") -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} EXCEPT ALL "
            f"SELECT * FROM {other._as_view()}")

    def intersectAll(self, other: "DataFrame") -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} INTERSECT ALL "
            f"SELECT * FROM {other._as_view()}")

    def subtract(self, other: "DataFrame") -> "DataFrame":
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} EXCEPT "
            f"SELECT * FROM {other._as_view()}")

    def dropDuplicates(self, subset=None) -> "DataFrame":
        if not subset:
            return self.distinct()
        cols = ", ".join(subset)
        all_cols = ", ".join(n for n, _t in self.plan.schema)
        return self.session.sql(
            f"SELECT {all_cols} FROM (SELECT *, row_number() OVER "
            f"(PARTITION BY {cols} ORDER BY {cols}) AS __rn FROM "
            f"{self._as_view()}) WHERE __rn = 1")

    drop_duplicates = dropDuplicates

    def sample(self, fraction: float, seed: Optional[int] = None,
               withReplacement: bool = False) -> "DataFrame":
        pct = float(fraction) * 100.0
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} TABLESAMPLE ({pct} PERCENT)"
            + (f" REPEATABLE ({seed})" if seed is not None else ""))

    def randomSplit(self, weights, seed: Optional[int] = None):
        total = float(sum(weights))
        bounds = []
        acc = 0.0
        for w in weights:
            acc += w / total
            bounds.append(acc)
        v = self._as_view()
        sd = seed if seed is not None else 42
        outs = []
        lo = 0.0
        for hi in bounds:
            outs.append(self.session.sql(
                f"SELECT * FROM (SELECT *, rand({sd}) AS __r FROM {v}) "
                f"WHERE __r >= {lo} AND __r < {hi}").drop("__r"))
            lo = hi
        return outs

    def repartition(self, n: int, *cols) -> "DataFrame":
        return self  # single-device partitioning is whole-column

    def coalesce(self, n: int) -> "DataFrame":
        return self

    def cache(self) -> "DataFrame":
        chunk = self.collect_chunk()
        src = S.ChunkSource(chunk=chunk, schema=self.plan.schema)
        src.schema = self.plan.schema
        return _MaterializedDataFrame(self.session, src, chunk)

    persist = cache

    def unpersist(self, blocking: bool = False) -> "DataFrame":
        return self

    def unpivot(self, ids, values, variableColumnName: str,
                valueColumnName: str) -> "DataFrame":
        idc = ", ".join(ids)
        vals = ", ".join(values)
        return self.session.sql(
            f"SELECT * FROM {self._as_view()} UNPIVOT "
            f"({valueColumnName} FOR {variableColumnName} IN ({vals}))")

    melt = unpivot

    @property
    def na(self):
        return _NaFunctions(self)

    @property
    def stat(self):
        return _StatFunctions(self)

    def foreach(self, fn):
        for row in self.collect():
            fn(row)

    def foreachPartition(self, fn):
        fn(iter(self.collect()))

    def rollup(self, *keys) -> "GroupedData":
        return GroupedData(self, list(keys), group_mode="ROLLUP")

    def cube(self, *keys) -> "GroupedData":
        return GroupedData(self, list(keys), group_mode="CUBE")


class _NaFunctions:
    """df.na.drop/fill/replace (PySpark surface over the engine fns)."""

    def __init__(self, df: "DataFrame"):
        self._df = df

    def drop(self, how: str = "any", subset=None):
        return self._df.dropna(how=how, subset=subset)

    def fill(self, value, subset=None):
        return self._df.fillna(value, subset=subset)

    def replace(self, to_replace, value=None, subset=None):
        return self._df.replace(to_replace, value, subset=subset)


class _StatFunctions:
    """df.stat.corr/cov/approxQuantile/crosstab/freqItems/sampleBy."""

    def __init__(self, df: "DataFrame"):
        self._df = df

    def corr(self, c1: str, c2: str) -> float:
        return self._df.session.sql(
            f"SELECT corr({c1}, {c2}) FROM {self._df._as_view()}"
        ).collect()[0][0]

    def cov(self, c1: str, c2: str) -> float:
        return self._df.session.sql(
            f"SELECT covar_samp({c1}, {c2}) FROM {self._df._as_view()}"
        ).collect()[0][0]

    def approxQuantile(self, col: str, probabilities, _relerr=0.0):
        v = self._df._as_view()
        sel = ", ".join(f"percentile({col}, {p})" for p in probabilities)
        row = self._df.session.sql(f"SELECT {sel} FROM {v}").collect()[0]
        return list(row)

    def crosstab(self, c1: str, c2: str) -> "DataFrame":
        v = self._df._as_view()
        vals = [r[0] for r in self._df.session.sql(
            f"SELECT DISTINCT {c2} FROM {v} ORDER BY 1").collect()]
        sel = ", ".join(
            f"count(CASE WHEN {c2} = '{x}' THEN 1 END) AS `{x}`"
            if isinstance(x, str) else
            f"count(CASE WHEN {c2} = {x} THEN 1 END) AS `{x}`"
            for x in vals)
        return self._df.session.sql(
            f"SELECT {c1} AS `{c1}_{c2}`{', ' if sel else ''}{sel} "
            f"FROM {v} GROUP BY {c1} ORDER BY 1")

    def freqItems(self, cols, support: float = 0.01):
        v = self._df._as_view()
        out = []
        n = max(self._df.count(), 1)
        for c in cols:
            rows = self._df.session.sql(
                f"SELECT {c}, count(*) AS c FROM {v} GROUP BY {c}"
            ).collect()
            out.append([r[0] for r in rows if r[1] / n >= support])
        return out

    def sampleBy(self, col: str, fractions: Dict, seed=None):
        v = self._df._as_view()
        sd = seed if seed is not None else 42
        conds = " OR ".join(
            (f"({col} = '{k}' AND rand({sd}) < {f})" if isinstance(k, str)
             else f"({col} = {k} AND rand({sd}) < {f})")
            for k, f in fractions.items())
        return self._df.session.sql(f"SELECT * FROM {v} WHERE {conds}")


class GroupedData:
    """df.group_by(...).agg(...) (SQL-backed); group_mode adds
    ROLLUP/CUBE (df.rollup/df.cube)."""

    def __init__(self, df: "DataFrame", keys, group_mode: str = ""):
        self._df = df
        self._keys = keys
        self._mode = group_mode

    def agg(self, *exprs) -> "DataFrame":
        view = self._df._as_view()
        sel = ", ".join(list(self._keys) + list(exprs))
        gb = ", ".join(self._keys) if self._keys else ""
        q = f"SELECT {sel} FROM {view}"
        if gb:
            if self._mode:
                q += f" GROUP BY {self._mode}({gb})"
            else:
                q += f" GROUP BY {gb}"
        return self._df.session.sql(q)

    def count(self) -> "DataFrame":
        return self.agg("count(*) AS count")

    def pivot(self, col: str, values=None) -> "_PivotedData":
        return _PivotedData(self._df, self._keys, col, values)


class _PivotedData:
    """df.groupBy(k).pivot(c, values).agg(expr) -> SQL PIVOT."""

    def __init__(self, df, keys, col, values):
        self._df = df
        self._keys = keys
        self._col = col
        self._values = values

    def agg(self, expr: str) -> "DataFrame":
        if not self._values:
            v = self._df._as_view()
            self._values = [r[0] for r in self._df.session.sql(
                f"SELECT DISTINCT {self._col} FROM {v} ORDER BY 1"
            ).collect()]
        vals = ", ".join(f"'{x}'" if isinstance(x, str) else str(x)
                         for x in self._values)
        v = self._df._as_view()
        keys = ", ".join(self._keys)
        inner_cols = ", ".join(
            [*self._keys, self._col] +
            _pivot_source_cols(expr))
        return self._df.session.sql(
            f"SELECT * FROM (SELECT {inner_cols} FROM {v}) "
            f"PIVOT ({expr} FOR {self._col} IN ({vals}))")


def _pivot_source_cols(expr: str):
    import re as _re

    inner = _re.findall(r"\(([^()]*)\)", expr)
    cols = []
    for grp in inner:
        for tok in _re.findall(r"[A-Za-z_][A-Za-z_0-9]*", grp):
            cols.append(tok)
    return cols or []


class _MaterializedDataFrame(DataFrame):
    def __init__(self, session, plan, chunk):
        super().__init__(session, plan)
        self._chunk = chunk

    def collect_chunk(self) -> Chunk:
        return self._chunk


class DataFrameReader:
    """spark.read.format(...).option(...).load(path) facade
    (ref: crates/sail-plan/src/resolver/query/read.rs data-source reads)."""

    def __init__(self, session: SessionContext):
        self._session = session
        self._format = "parquet"
        self._options: Dict[str, str] = {}
        self._schema = None

    def format(self, fmt: str) -> "DataFrameReader":
        self._format = fmt
        return self

    def option(self, k: str, v) -> "DataFrameReader":
        self._options[k] = str(v)
        return self

    def options(self, **kw) -> "DataFrameReader":
        for k, v in kw.items():
            self._options[k] = str(v)
        return self

    def schema(self, schema) -> "DataFrameReader":
        self._schema = schema
        return self

    def load(self, path) -> "DataFrame":
        paths = path if isinstance(path, list) else [path]
        plan = S.DataSourceRead(format=self._format, paths=paths,
                                options=dict(self._options), user_schema=self._schema)
        resolved = _CatalogAdapter(self._session).resolve(plan)
        return DataFrame(self._session, resolved)

    def parquet(self, *paths) -> "DataFrame":
        return self.format("parquet").load(list(paths))

    def csv(self, *paths, **kw) -> "DataFrame":
        return self.format("csv").options(**kw).load(list(paths))

    def json(self, *paths) -> "DataFrame":
        return self.format("json").load(list(paths))

    def table(self, name: str) -> "DataFrame":
        return self._session.table(name)


class DataFrameWriter:
    """df.write.format(...).mode(...).save(path)."""

    def __init__(self, df: "DataFrame"):
        self._df = df
        self._format = "parquet"
        self._mode = "error"
        self._options: Dict[str, str] = {}
        self._partition_by: List[str] = []

    def format(self, fmt: str) -> "DataFrameWriter":
        self._format = fmt
        return self

    def mode(self, m: str) -> "DataFrameWriter":
        self._mode = {"errorifexists": "error", "default": "error"}.get(m.lower(), m.lower())
        return self

    def option(self, k: str, v) -> "DataFrameWriter":
        self._options[k] = str(v)
        return self

    def partitionBy(self, *cols) -> "DataFrameWriter":
        self._partition_by = list(cols)
        return self

    def save(self, path: str):
        from ..datasource.registry import write_source

        chunk = self._df.collect_chunk()
        write_source(self._format, path, chunk, self._mode, self._options,
                     self._partition_by)

    def parquet(self, path: str):
        self.format("parquet").save(path)

    def csv(self, path: str):
        self.format("csv").save(path)

    def saveAsTable(self, name: str):
        chunk = self._df.collect_chunk()
        self._df.session.catalog.register_table_chunk(
            name, chunk, self._df.plan.schema)
