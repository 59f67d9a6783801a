"""Spark Connect Relation/Expression proto -> unresolved plan spec.

The reference converts the full Relation tree in
crates/sail-spark-connect/src/proto/plan.rs (2.3k LoC) + proto/expression.rs;
this module covers the core DataFrame-API surface so an unmodified PySpark
client's non-SQL plans execute: Read (named table + data source), Project,
Filter, Join, SetOperation, Sort, Limit/Offset/Tail, Aggregate (groupby),
LocalRelation (Arrow IPC), Range, SubqueryAlias, ToDF, WithColumns,
WithColumnsRenamed, Drop, Deduplicate, Sample, ShowString, Hint, NAFill/
NADrop, Repartition (no-op single-plan engine).

Field numbers follow the public Apache Spark `spark/connect/*.proto`
definitions (relations.proto / expressions.proto)."""
from __future__ import annotations

from typing import List, Optional

from ..engine import types as T
from ..plan import spec as S
from . import wire as W


class Unsupported(Exception):
    pass


# Relation oneof field numbers (relations.proto)
REL_READ = 2
REL_PROJECT = 3
REL_FILTER = 4
REL_JOIN = 5
REL_SET_OP = 6
REL_SORT = 7
REL_LIMIT = 8
REL_AGGREGATE = 9
REL_SQL = 10
REL_LOCAL = 11
REL_SAMPLE = 12
REL_OFFSET = 13
REL_DEDUP = 14
REL_RANGE = 15
REL_SUBQUERY_ALIAS = 16
REL_REPARTITION = 17
REL_TO_DF = 18
REL_WITH_COLS_RENAMED = 19
REL_SHOW_STRING = 20
REL_DROP = 21
REL_TAIL = 22
REL_WITH_COLUMNS = 23
REL_HINT = 24
REL_FILL_NA = 90
REL_DROP_NA = 91

# Expression oneof field numbers (expressions.proto)
EX_LITERAL = 1
EX_ATTR = 2
EX_FUNC = 3
EX_EXPR_STRING = 4
EX_STAR = 5
EX_ALIAS = 6
EX_CAST = 7
EX_SORT_ORDER = 9
EX_EXTRACT = 12

# Literal oneof
LIT_NULL = 1
LIT_BINARY = 2
LIT_BOOL = 3
LIT_BYTE = 4
LIT_SHORT = 5
LIT_INT = 6
LIT_LONG = 7
LIT_FLOAT = 10
LIT_DOUBLE = 11
LIT_DECIMAL = 12
LIT_STRING = 13
LIT_DATE = 16
LIT_TIMESTAMP = 17
LIT_TIMESTAMP_NTZ = 18

_JOIN_TYPES = {1: "inner", 2: "full", 3: "left", 4: "right", 5: "anti",
               6: "semi", 7: "cross"}
_SET_OPS = {1: "intersect", 2: "union", 3: "except"}


def _is_agg(name: str) -> bool:
    from ..engine.aggregates import UDAFS
    from ..functions.registry import AGG_FUNCTIONS

    return name in AGG_FUNCTIONS or name in UDAFS


def _fixed32(raw: int) -> float:
    import struct

    return struct.unpack("<f", struct.pack("<I", raw & 0xFFFFFFFF))[0]


def _fixed64(raw: int) -> float:
    import struct

    return struct.unpack("<d", struct.pack("<Q", raw & 0xFFFFFFFFFFFFFFFF))[0]


def _signed(v: int) -> int:
    """Protobuf int32/int64 arrive as raw varints; two's-complement them."""
    if v >= (1 << 63):
        v -= 1 << 64
    return v


def decode_literal(buf: bytes) -> S.Literal:
    f = W.parse(buf)
    if LIT_NULL in f:
        return S.Literal(None)
    if LIT_BOOL in f:
        return S.Literal(bool(W.first_varint(f, LIT_BOOL)))
    for num in (LIT_BYTE, LIT_SHORT, LIT_INT, LIT_LONG):
        if num in f:
            return S.Literal(_signed(W.first_varint(f, num)))
    if LIT_FLOAT in f:
        return S.Literal(_fixed32(W.first_varint(f, LIT_FLOAT)), dtype=T.F32)
    if LIT_DOUBLE in f:
        return S.Literal(_fixed64(W.first_varint(f, LIT_DOUBLE)))
    if LIT_DECIMAL in f:
        d = W.parse(W.first(f, LIT_DECIMAL))
        from decimal import Decimal

        return S.Literal(Decimal(W.first_str(d, 1)))
    if LIT_STRING in f:
        return S.Literal(W.first_str(f, LIT_STRING))
    if LIT_BINARY in f:
        return S.Literal(W.first(f, LIT_BINARY))
    if LIT_DATE in f:
        import datetime

        days = _signed(W.first_varint(f, LIT_DATE))
        return S.Literal(datetime.date(1970, 1, 1)
                         + datetime.timedelta(days=days), dtype=T.DATE)
    for num in (LIT_TIMESTAMP, LIT_TIMESTAMP_NTZ):
        if num in f:
            return S.Literal(_signed(W.first_varint(f, num)),
                             dtype=T.TIMESTAMP)
    raise Unsupported(f"literal fields {sorted(f)}")


def decode_expr(buf: bytes) -> S.Expr:
    f = W.parse(buf)
    if EX_LITERAL in f:
        return decode_literal(W.first(f, EX_LITERAL))
    if EX_ATTR in f:
        a = W.parse(W.first(f, EX_ATTR))
        name = W.first_str(a, 1)
        if "." in name:
            q, _, base = name.rpartition(".")
            return S.Col(base, qualifier=q)
        return S.Col(name)
    if EX_FUNC in f:
        fn = W.parse(W.first(f, EX_FUNC))
        name = W.first_str(fn, 1).lower()
        args = [decode_expr(a) for a in fn.get(2, [])]
        distinct = bool(W.first_varint(fn, 3))
        # PySpark Column operators arrive as functions named by symbol
        binop = {"==": "=", "=": "=", "!=": "!=", "<>": "!=", "<": "<",
                 "<=": "<=", ">": ">", ">=": ">=", "+": "+", "-": "-",
                 "*": "*", "/": "/", "%": "%", "and": "and", "or": "or",
                 "&": "&", "|": "|", "^": "^"}.get(name)
        if binop is not None and len(args) == 2:
            return S.BinaryOp(op=binop, left=args[0], right=args[1])
        if name in ("not", "!") and len(args) == 1:
            return S.UnaryOp(op="not", child=args[0])
        if name in ("negative", "negate") and len(args) == 1:
            return S.UnaryOp(op="neg", child=args[0])
        if name == "-" and len(args) == 1:
            return S.UnaryOp(op="neg", child=args[0])
        if name == "isnull" and len(args) == 1:
            return S.Func("isnull", args)
        if _is_agg(name):
            return S.AggFunc(name, args, distinct=distinct)
        return S.Func(name, args)
    if EX_EXPR_STRING in f:
        from ..sql.parser import Parser

        text = W.first_str(W.parse(W.first(f, EX_EXPR_STRING)), 1)
        return Parser(text).parse_expr()
    if EX_STAR in f:
        st = W.parse(W.first(f, EX_STAR))
        target = W.first_str(st, 1)
        qual = target[:-2] if target.endswith(".*") else None
        return S.Star(qualifier=qual) if qual else S.Star()
    if EX_ALIAS in f:
        al = W.parse(W.first(f, EX_ALIAS))
        child = decode_expr(W.first(al, 1, b""))
        names = [n.decode() for n in al.get(2, [])]
        if len(names) != 1:
            raise Unsupported("multi-name alias")
        return S.Alias(child, names[0])
    if EX_CAST in f:
        c = W.parse(W.first(f, EX_CAST))
        child = decode_expr(W.first(c, 1, b""))
        ts = W.first_str(c, 3)
        if not ts:
            raise Unsupported("cast with DataType proto (send type_str)")
        return S.Cast(child, T.type_from_name(ts))
    if EX_SORT_ORDER in f:
        return decode_sort_order(W.first(f, EX_SORT_ORDER))
    if EX_EXTRACT in f:
        e = W.parse(W.first(f, EX_EXTRACT))
        child = decode_expr(W.first(e, 1, b""))
        extraction = decode_expr(W.first(e, 2, b""))
        return S.Func("get", [child, extraction])
    raise Unsupported(f"expression fields {sorted(f)}")


def decode_sort_order(buf: bytes) -> S.SortKey:
    so = W.parse(buf)
    child = decode_expr(W.first(so, 1, b""))
    asc = W.first_varint(so, 2, 1) != 2
    null_ord = W.first_varint(so, 3, 0)
    nulls_first = {0: None, 1: True, 2: False}[null_ord]
    return S.SortKey(child, asc, nulls_first)


class RelationConverter:
    """One Relation tree -> unresolved spec plan (resolved by the session's
    normal resolve/optimize pipeline, same as SQL)."""

    def __init__(self, session):
        self.session = session

    def convert(self, rel_bytes: bytes) -> S.Plan:
        r = W.parse(rel_bytes)
        if REL_SQL in r:
            sql = W.first_str(W.parse(W.first(r, REL_SQL)), 1)
            return self.session.parse(sql)
        if REL_READ in r:
            return self._read(W.parse(W.first(r, REL_READ)))
        if REL_PROJECT in r:
            p = W.parse(W.first(r, REL_PROJECT))
            inp = self._input(p, 1)
            exprs = [decode_expr(e) for e in p.get(3, [])]
            return S.Project(input=inp, exprs=exprs or [S.Star()])
        if REL_FILTER in r:
            p = W.parse(W.first(r, REL_FILTER))
            return S.Filter(input=self._input(p, 1),
                            condition=decode_expr(W.first(p, 2, b"")))
        if REL_JOIN in r:
            return self._join(W.parse(W.first(r, REL_JOIN)))
        if REL_SET_OP in r:
            p = W.parse(W.first(r, REL_SET_OP))
            op = _SET_OPS.get(W.first_varint(p, 3), None)
            if op is None:
                raise Unsupported("set op type")
            return S.SetOp(op=op, left=self._input(p, 1),
                           right=self._input(p, 2),
                           is_all=bool(W.first_varint(p, 4)),
                           by_name=bool(W.first_varint(p, 5)))
        if REL_SORT in r:
            p = W.parse(W.first(r, REL_SORT))
            keys = [decode_sort_order(o) for o in p.get(2, [])]
            return S.Sort(input=self._input(p, 1), keys=keys)
        if REL_LIMIT in r:
            p = W.parse(W.first(r, REL_LIMIT))
            return S.Limit(input=self._input(p, 1),
                           n=W.first_varint(p, 2))
        if REL_OFFSET in r:
            p = W.parse(W.first(r, REL_OFFSET))
            return S.Limit(input=self._input(p, 1), n=None,
                           offset=W.first_varint(p, 2))
        if REL_TAIL in r:
            p = W.parse(W.first(r, REL_TAIL))
            n = W.first_varint(p, 2)
            inp = self._input(p, 1)
            # tail is an action in PySpark: execute eagerly, keep last n
            plan = self.session.optimize(self.session.resolve(inp))
            chunk = self.session.execute_plan(plan)
            total = chunk.num_rows
            lo = max(total - n, 0)
            from ..engine.chunk import Chunk

            sliced = Chunk([c.slice(lo, total - lo) for c in chunk.columns],
                           list(chunk.names))
            node = S.ChunkSource(chunk=sliced, schema=list(plan.schema))
            return node
        if REL_AGGREGATE in r:
            return self._aggregate(W.parse(W.first(r, REL_AGGREGATE)))
        if REL_LOCAL in r:
            return self._local(W.parse(W.first(r, REL_LOCAL)))
        if REL_RANGE in r:
            p = W.parse(W.first(r, REL_RANGE))
            start = _signed(W.first_varint(p, 1, 0))
            end = _signed(W.first_varint(p, 2, 0))
            step = _signed(W.first_varint(p, 3, 1)) or 1
            return S.Range(start=start, end=end, step=step)
        if REL_SUBQUERY_ALIAS in r:
            p = W.parse(W.first(r, REL_SUBQUERY_ALIAS))
            return S.SubqueryAlias(input=self._input(p, 1),
                                   alias=W.first_str(p, 2))
        if REL_REPARTITION in r:
            p = W.parse(W.first(r, REL_REPARTITION))
            return self._input(p, 1)  # single-plan SPMD: repartition no-op
        if REL_TO_DF in r:
            p = W.parse(W.first(r, REL_TO_DF))
            names = [n.decode() for n in p.get(2, [])]
            sub = S.SubqueryAlias(input=self._input(p, 1), alias="__todf__",
                                  column_aliases=names)
            return sub
        if REL_WITH_COLS_RENAMED in r:
            p = W.parse(W.first(r, REL_WITH_COLS_RENAMED))
            renames = {}
            for rn in p.get(3, []):
                rf = W.parse(rn)
                renames[W.first_str(rf, 1).lower()] = W.first_str(rf, 2)
            for key in p.get(2, []):  # legacy map<string,string> form
                kvf = W.parse(key)
                renames[W.first_str(kvf, 1).lower()] = W.first_str(kvf, 2)
            node = self._input(p, 1)
            names = self._schema_names(node)
            exprs = [S.Alias(S.Col(n), renames[n.lower()])
                     if n.lower() in renames else S.Col(n) for n in names]
            return S.Project(input=node, exprs=exprs)
        if REL_WITH_COLUMNS in r:
            p = W.parse(W.first(r, REL_WITH_COLUMNS))
            aliases = []
            for al in p.get(2, []):
                af = W.parse(al)
                child = decode_expr(W.first(af, 1, b""))
                names = [n.decode() for n in af.get(2, [])]
                aliases.append(S.Alias(child, names[0]))
            node = self._input(p, 1)
            names = self._schema_names(node)
            by_name = {a.name.lower(): a for a in aliases}
            exprs = []
            for n in names:  # replace in place, Spark withColumn semantics
                exprs.append(by_name.pop(n.lower(), None) or S.Col(n))
            exprs.extend(by_name.values())  # brand-new columns at the end
            return S.Project(input=node, exprs=exprs)
        if REL_DROP in r:
            p = W.parse(W.first(r, REL_DROP))
            drop = {n.decode().lower() for n in p.get(3, [])}
            for e in p.get(2, []):
                ex = decode_expr(e)
                if isinstance(ex, S.Col):
                    drop.add(ex.name.lower())
            node = self._input(p, 1)
            keep = [n for n in self._schema_names(node)
                    if n.lower() not in drop]
            return S.Project(input=node, exprs=[S.Col(n) for n in keep])
        if REL_DEDUP in r:
            p = W.parse(W.first(r, REL_DEDUP))
            names = {n.decode().lower() for n in p.get(2, [])}
            inp = self._input(p, 1)
            if not names or W.first_varint(p, 3):
                return S.Distinct(input=inp)
            # dropDuplicates(subset): group by the subset, keep an arbitrary
            # row's other columns (Spark keeps "first" nondeterministically)
            all_names = self._schema_names(inp)
            group = [S.Col(n) for n in all_names if n.lower() in names]
            aggs = [S.Col(n) if n.lower() in names
                    else S.Alias(S.AggFunc("first", [S.Col(n)]), n)
                    for n in all_names]
            return S.Aggregate(input=inp, group_by=group, aggs=aggs)
        if REL_SAMPLE in r:
            p = W.parse(W.first(r, REL_SAMPLE))
            lo = _fixed64(W.first_varint(p, 2, 0))
            hi = _fixed64(W.first_varint(p, 3, 0))
            seed = _signed(W.first_varint(p, 5, 0)) or None
            return S.Sample(input=self._input(p, 1), fraction=hi - lo,
                            seed=seed)
        if REL_SHOW_STRING in r:
            p = W.parse(W.first(r, REL_SHOW_STRING))
            return self._show_string(p)
        if REL_HINT in r:
            p = W.parse(W.first(r, REL_HINT))
            return self._input(p, 1)
        if REL_FILL_NA in r or REL_DROP_NA in r:
            raise Unsupported("fillna/dropna over the wire (use SQL)")
        raise Unsupported(f"relation fields {sorted(r)}")

    # -- helpers -----------------------------------------------------------
    def _schema_names(self, node: S.Plan) -> List[str]:
        """Column names of a subtree: resolve a throwaway copy (the original
        unresolved tree is resolved once, later, by the normal pipeline)."""
        import copy

        probe = self.session.resolve(copy.deepcopy(node)) \
            if node.schema is None else node
        return [n for n, _ in probe.schema]

    def _input(self, fields, num) -> S.Plan:
        sub = W.first(fields, num)
        if sub is None:
            raise Unsupported("missing input relation")
        return self.convert(sub)

    def _read(self, p) -> S.Plan:
        nt = W.first(p, 1)
        ds = W.first(p, 2)
        if nt is not None:
            ntf = W.parse(nt)
            return S.Read(table=W.first_str(ntf, 1))
        if ds is not None:
            df = W.parse(ds)
            fmt = W.first_str(df, 1) or "parquet"
            paths = [x.decode() for x in df.get(4, [])]
            options = {}
            for kv in df.get(3, []):
                kvf = W.parse(kv)
                options[W.first_str(kvf, 1)] = W.first_str(kvf, 2)
            return S.DataSourceRead(format=fmt, paths=paths, options=options)
        raise Unsupported("read without table or source")

    def _join(self, p) -> S.Plan:
        how = _JOIN_TYPES.get(W.first_varint(p, 4), "inner")
        cond = W.first(p, 3)
        using = [c.decode() for c in p.get(5, [])]
        return S.Join(left=self._input(p, 1), right=self._input(p, 2),
                      how=how,
                      on=decode_expr(cond) if cond is not None else None,
                      using=using or None)

    def _aggregate(self, p) -> S.Plan:
        # GroupType: 1=GROUPBY 2=ROLLUP 3=CUBE 4=PIVOT 5=GROUPING_SETS
        gtype = W.first_varint(p, 2, 1)
        group = [decode_expr(e) for e in p.get(3, [])]
        aggs = [decode_expr(e) for e in p.get(4, [])]
        inp = self._input(p, 1)
        if gtype == 4:
            pv = W.first(p, 5)
            if pv is None:
                raise Unsupported("PIVOT group type without pivot info")
            pf = W.parse(pv)
            col = decode_expr(W.first(pf, 1))
            vals = [decode_literal(v) for v in pf.get(2, [])]
            if not vals:
                raise Unsupported("pivot without explicit values")
            if len(aggs) != 1:
                raise Unsupported("pivot requires exactly one aggregate")
            return S.Pivot(input=inp, agg=aggs[0], pivot=col, values=vals)
        gsets = None
        if gtype == 2:   # ROLLUP: prefixes, longest first, down to ()
            gsets = [list(range(k)) for k in range(len(group), -1, -1)]
        elif gtype == 3:  # CUBE: every subset
            import itertools as _it

            gsets = [list(c) for k in range(len(group), -1, -1)
                     for c in _it.combinations(range(len(group)), k)]
        elif gtype == 5:  # explicit GROUPING SETS: match exprs by shape
            reprs = [repr(g) for g in group]
            gsets = []
            for gs in p.get(6, []):
                gf = W.parse(gs)
                idxs = []
                for ge in gf.get(1, []):
                    r = repr(decode_expr(ge))
                    if r not in reprs:
                        raise Unsupported(
                            "grouping set expr not among group columns")
                    idxs.append(reprs.index(r))
                gsets.append(idxs)
        elif gtype not in (0, 1):
            raise Unsupported(f"aggregate group type {gtype}")
        # Spark: output columns = grouping expressions ++ aggregate exprs
        return S.Aggregate(input=inp, group_by=list(group),
                           aggs=list(group) + aggs, grouping_sets=gsets)

    def _local(self, p) -> S.Plan:
        data = W.first(p, 1)
        if data is None:
            raise Unsupported("LocalRelation without data")
        import io

        import pyarrow as pa

        from ..datasource.arrow_io import arrow_to_table
        from ..engine.chunk import Chunk

        with pa.ipc.open_stream(io.BytesIO(data)) as rd:
            tbl = rd.read_all()
        t = arrow_to_table(tbl, device=self.session.device)
        chunk = Chunk.from_table(t)
        node = S.ChunkSource(chunk=chunk,
                             schema=[(n, c.dtype)
                                     for n, c in t.columns.items()])
        return node

    def _show_string(self, p) -> S.Plan:
        num_rows = W.first_varint(p, 2, 20)
        truncate = W.first_varint(p, 3, 20)
        inp = self._input(p, 1)
        plan = self.session.optimize(self.session.resolve(inp))
        chunk = self.session.execute_plan(
            S.Limit(input=plan, n=num_rows + 1, schema=plan.schema))
        names = [n for n, _ in plan.schema]
        cols = [c.to_pylist() for c in chunk.columns]
        rows = list(zip(*cols))[:num_rows] if cols else []

        def cell(v):
            s = "NULL" if v is None else str(v)
            if truncate and len(s) > truncate:
                s = s[: truncate - 3] + "..."
            return s

        table = [names] + [[cell(v) for v in row] for row in rows]
        widths = [max(len(r[i]) for r in table) for i in range(len(names))]
        sep = "+" + "+".join("-" * (w + 2) for w in widths) + "+"
        lines = [sep,
                 "|" + "|".join(f" {names[i]:<{widths[i]}} "
                                for i in range(len(names))) + "|",
                 sep]
        for row in table[1:]:
            lines.append("|" + "|".join(
                f" {row[i]:<{widths[i]}} " for i in range(len(names))) + "|")
        lines.append(sep)
        if chunk.num_rows > num_rows:
            lines.append(f"only showing top {num_rows} rows")
        text = "\n".join(lines) + "\n"
        return S.LocalRelation(data={"show_string": [text]},
                               schema=[("show_string", T.STRING)])
