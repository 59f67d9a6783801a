"""Spark SQL parser.

Hand-written recursive-descent statement/query parser with Pratt expression
parsing — same layering as the reference's parser crate
(ref: crates/sail-sql-parser/src/parser.rs:19 statement/query/expression
parsers), re-implemented from the Spark SQL grammar rather than ported.

Produces the spec IR of plan/spec.py.
"""
from __future__ import annotations

import datetime as _dt
from typing import List, Optional, Tuple

from ..engine import types as T
from ..plan import spec as S
from .lexer import SqlError, Token, tokenize

# Keywords that terminate expression parsing contexts
_RESERVED_STOP = {
    "FROM", "WHERE", "GROUP", "HAVING", "ORDER", "LIMIT", "OFFSET", "UNION",
    "INTERSECT", "EXCEPT", "MINUS", "JOIN", "INNER", "LEFT", "RIGHT", "FULL",
    "CROSS", "ON", "USING", "AS", "WHEN", "THEN", "ELSE", "END", "AND", "OR",
    "ASC", "DESC", "NULLS", "BY", "WITH", "SELECT", "DISTINCT", "ALL",
    "SEMI", "ANTI", "NATURAL", "LATERAL", "WINDOW", "CLUSTER", "DISTRIBUTE",
    "SORT", "OVER", "ROWS", "RANGE", "PARTITION", "FOR", "CASE", "INTO",
    "TABLESAMPLE", "PIVOT", "UNPIVOT",
}

_JOIN_TYPES = {
    "INNER": "inner", "LEFT": "left", "RIGHT": "right", "FULL": "full",
    "CROSS": "cross", "SEMI": "semi", "ANTI": "anti",
}


class Parser:
    def __init__(self, sql: str):
        self.sql = sql
        self.toks = tokenize(sql)
        self.i = 0

    # -- token helpers -----------------------------------------------------
    def peek(self, off: int = 0) -> Token:
        return self.toks[min(self.i + off, len(self.toks) - 1)]

    def next(self) -> Token:
        t = self.toks[self.i]
        if t.kind != "eof":
            self.i += 1
        return t

    def at_kw(self, *kws: str) -> bool:
        t = self.peek()
        return t.kind == "ident" and t.upper in kws

    def eat_kw(self, *kws: str) -> bool:
        if self.at_kw(*kws):
            self.next()
            return True
        return False

    def expect_kw(self, kw: str):
        if not self.eat_kw(kw):
            raise SqlError(f"expected {kw}, found {self.peek().value!r}", self.sql, self.peek().pos)

    def at_op(self, *ops: str) -> bool:
        t = self.peek()
        return t.kind == "op" and t.value in ops

    def eat_op(self, *ops: str) -> bool:
        if self.at_op(*ops):
            self.next()
            return True
        return False

    def expect_op(self, op: str):
        if not self.eat_op(op):
            raise SqlError(f"expected {op!r}, found {self.peek().value!r}", self.sql, self.peek().pos)

    def ident(self) -> str:
        t = self.peek()
        if t.kind != "ident":
            raise SqlError(f"expected identifier, found {t.value!r}", self.sql, t.pos)
        self.next()
        return t.value

    # =======================================================================
    # Statements
    # =======================================================================
    def parse_statements(self) -> List[S.Plan]:
        out = []
        while self.peek().kind != "eof":
            out.append(self.parse_statement())
            while self.eat_op(";"):
                pass
        return out

    def parse_statement(self) -> S.Plan:
        if self.at_kw("SELECT", "WITH", "VALUES", "TABLE") or self.at_op("("):
            return self.parse_query()
        if self.at_kw("EXPLAIN"):
            self.next()
            mode = "simple"
            if self.at_kw("EXTENDED", "CODEGEN", "COST", "FORMATTED", "ANALYZE"):
                mode = self.next().value.lower()
            return S.Explain(input=self.parse_statement(), mode=mode)
        if self.at_kw("SET"):
            return self._parse_set()
        if self.at_kw("CREATE"):
            return self._parse_create()
        if self.at_kw("DROP"):
            return self._parse_drop()
        if self.at_kw("INSERT"):
            return self._parse_insert()
        if self.at_kw("MERGE"):
            return self._parse_merge()
        if self.at_kw("UPDATE"):
            return self._parse_update()
        if self.at_kw("DELETE"):
            return self._parse_delete()
        if self.at_kw("SHOW"):
            self.next()
            if self.eat_kw("FUNCTIONS"):
                pattern = None
                if self.eat_kw("LIKE"):
                    pattern = self.next().value
                return S.ShowFunctions(pattern=pattern)
            if self.eat_kw("DATABASES") or self.eat_kw("SCHEMAS"):
                return S.ShowDatabases()
            if self.eat_kw("CATALOGS"):
                return S.ShowCatalogs()
            if self.eat_kw("COLUMNS"):
                self.eat_kw("IN") or self.eat_kw("FROM")
                return S.ShowColumns(name=self._qualified_name())
            if self.eat_kw("CREATE"):
                self.expect_kw("TABLE")
                return S.ShowCreateTable(name=self._qualified_name())
            if self.eat_kw("VIEWS"):
                self.eat_kw("IN") and self.ident()
                pattern = None
                if self.eat_kw("LIKE"):
                    pattern = self.next().value
                return S.ShowViews(pattern=pattern)
            if self.eat_kw("PARTITIONS"):
                return S.ShowPartitions(name=self._qualified_name())
            if self.eat_kw("TBLPROPERTIES"):
                return S.ShowTblProperties(name=self._qualified_name())
            self.expect_kw("TABLES")
            self.eat_kw("IN") and self.ident()
            pattern = None
            if self.eat_kw("LIKE"):
                pattern = self.next().value
            return S.ShowTables(pattern=pattern)
        if self.at_kw("USE"):
            self.next()
            self.eat_kw("DATABASE") or self.eat_kw("SCHEMA")
            return S.UseDatabase(name=self.ident())
        if self.at_kw("REFRESH"):
            self.next()
            self.eat_kw("TABLE")
            return S.RefreshTable(name=self._qualified_name())
        if self.at_kw("TRUNCATE"):
            self.next()
            self.eat_kw("TABLE")
            return S.TruncateTable(name=self._qualified_name())
        if self.at_kw("COMMENT"):
            self.next()
            self.expect_kw("ON")
            self.expect_kw("TABLE")
            name = self._qualified_name()
            self.expect_kw("IS")
            if self.eat_kw("NULL"):
                c = None
            else:
                c = self.next().value
            return S.CommentOn(kind="table", name=name, comment=c)
        if self.at_kw("ALTER"):
            self.next()
            self.expect_kw("TABLE")
            name = self._qualified_name()
            if self.eat_kw("ADD"):
                self.eat_kw("COLUMNS") or self.eat_kw("COLUMN")
                cols = []
                paren = self.eat_op("(")
                while True:
                    cn = self.ident()
                    ct = self._parse_type()
                    cols.append((cn, ct))
                    if not self.eat_op(","):
                        break
                if paren:
                    self.expect_op(")")
                return S.AlterTable(name=name, action="add_columns", columns=cols)
            if self.eat_kw("DROP"):
                self.eat_kw("COLUMNS") or self.eat_kw("COLUMN")
                paren = self.eat_op("(")
                col = self.ident()
                if paren:
                    self.expect_op(")")
                return S.AlterTable(name=name, action="drop_column", column=col)
            if self.eat_kw("RENAME"):
                if self.eat_kw("TO"):
                    return S.AlterTable(name=name, action="rename_table",
                                        new_name=self._qualified_name())
                self.expect_kw("COLUMN")
                col = self.ident()
                self.expect_kw("TO")
                return S.AlterTable(name=name, action="rename_column",
                                    column=col, new_name=self.ident())
            raise SqlError("unsupported ALTER TABLE action", self.sql, self.peek().pos)
        if self.at_kw("CACHE"):
            self.next()
            self.eat_kw("LAZY")
            self.expect_kw("TABLE")
            name = self._qualified_name()
            inp = None
            if self.eat_kw("AS"):
                inp = self.parse_query()
            return S.CacheTable(name=name, input=inp)
        if self.at_kw("UNCACHE"):
            self.next()
            self.expect_kw("TABLE")
            self.eat_kw("IF") and self.expect_kw("EXISTS")
            return S.UncacheTable(name=self._qualified_name())
        if self.at_kw("ANALYZE"):
            self.next()
            self.expect_kw("TABLE")
            name = self._qualified_name()
            self.expect_kw("COMPUTE")
            self.expect_kw("STATISTICS")
            cols = None
            if self.eat_kw("FOR"):
                if self.eat_kw("ALL"):
                    self.expect_kw("COLUMNS")
                    cols = []
                else:
                    self.expect_kw("COLUMNS")
                    cols = [self.ident()]
                    while self.eat_op(","):
                        cols.append(self.ident())
            return S.AnalyzeTable(name=name, columns=cols)
        if self.at_kw("VACUUM"):
            self.next()
            name = self._qualified_name()
            hours = None
            if self.eat_kw("RETAIN"):
                t = self.next()
                hours = float(t.value.split("#")[0])
                self.eat_kw("HOURS")
            dry = bool(self.eat_kw("DRY")) and bool(self.eat_kw("RUN"))
            return S.VacuumTable(name=name, retention_hours=hours, dry_run=dry)
        if self.at_kw("DESCRIBE", "DESC"):
            self.next()
            if self.eat_kw("HISTORY"):
                return S.DescribeHistory(name=self._qualified_name())
            if self.eat_kw("QUERY") or self.at_kw("SELECT", "WITH",
                                                  "VALUES"):
                return S.DescribeQuery(input=self.parse_query())
            self.eat_kw("TABLE")
            self.eat_kw("EXTENDED")
            return S.DescribeTable(name=self._qualified_name())
        raise SqlError(f"unsupported statement start {self.peek().value!r}", self.sql, self.peek().pos)

    def _parse_set(self) -> S.Plan:
        self.expect_kw("SET")
        # SET key = value (key may be dotted idents)
        parts = [self.ident()]
        while self.eat_op("."):
            parts.append(self.ident())
        key = ".".join(parts)
        value = None
        if self.eat_op("="):
            vt = self.next()
            value = vt.value
            # values may continue as dotted/word tokens until ; or eof
            while self.peek().kind in ("ident", "number") or self.at_op("."):
                value += self.next().value
        return S.SetConfig(key=key, value=value)

    def _parse_create(self) -> S.Plan:
        self.expect_kw("CREATE")
        replace = False
        if self.eat_kw("OR"):
            self.expect_kw("REPLACE")
            replace = True
        if self.at_kw("DATABASE", "SCHEMA"):
            self.next()
            ine = self._if_not_exists()
            name = self.ident()
            comment = ""
            if self.eat_kw("COMMENT"):
                comment = self.next().value
            return S.CreateDatabase(name=name, if_not_exists=ine,
                                    comment=comment)
        temp = self.eat_kw("TEMP") or self.eat_kw("TEMPORARY")
        if self.eat_kw("VIEW"):
            if_not_exists = self._if_not_exists()
            name = self._qualified_name()
            self.expect_kw("AS")
            return S.CreateView(name=name, input=self.parse_query(), replace=replace or if_not_exists,
                                temporary=True if temp else True)
        if self.eat_kw("TABLE"):
            if_not_exists = self._if_not_exists()
            name = self._qualified_name()
            columns: List[Tuple[str, T.DataType]] = []
            fmt, location = None, None
            options = {}
            if self.at_op("(") and not self._peek_is_query_paren():
                self.expect_op("(")
                while True:
                    cname = self.ident()
                    ctype = self._parse_type()
                    # ignore column constraints (NOT NULL etc.)
                    while not self.at_op(",") and not self.at_op(")"):
                        self.next()
                    columns.append((cname, ctype))
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
            if self.eat_kw("USING"):
                fmt = self.ident().lower()
            if self.eat_kw("STORED"):
                self.expect_kw("AS")
                fmt = self.ident().lower()
            if self.eat_kw("OPTIONS") or self.eat_kw("TBLPROPERTIES"):
                self.expect_op("(")
                while not self.at_op(")"):
                    k = self.next().value
                    self.eat_op("=")
                    v = self.next().value
                    options[k] = v
                    self.eat_op(",")
                self.expect_op(")")
            if self.eat_kw("LOCATION"):
                location = self.next().value
            inp = None
            if self.eat_kw("AS"):
                inp = self.parse_query()
            return S.CreateTable(name=name, columns=columns, input=inp, format=fmt,
                                 location=location, replace=replace, if_not_exists=if_not_exists,
                                 options=options)
        raise SqlError("expected VIEW or TABLE after CREATE", self.sql, self.peek().pos)

    def _if_not_exists(self) -> bool:
        if self.eat_kw("IF"):
            self.expect_kw("NOT")
            self.expect_kw("EXISTS")
            return True
        return False

    def _parse_drop(self) -> S.Plan:
        self.expect_kw("DROP")
        if self.at_kw("DATABASE", "SCHEMA"):
            self.next()
            if_exists = False
            if self.eat_kw("IF"):
                self.expect_kw("EXISTS")
                if_exists = True
            name = self.ident()
            cascade = self.eat_kw("CASCADE")
            self.eat_kw("RESTRICT")
            return S.DropDatabase(name=name, if_exists=if_exists,
                                  cascade=cascade)
        is_view = self.eat_kw("VIEW")
        if not is_view:
            self.expect_kw("TABLE")
        if_exists = False
        if self.eat_kw("IF"):
            self.expect_kw("EXISTS")
            if_exists = True
        return S.DropTable(name=self._qualified_name(), if_exists=if_exists, is_view=is_view)

    def _parse_insert(self) -> S.Plan:
        self.expect_kw("INSERT")
        overwrite = self.eat_kw("OVERWRITE")
        if not overwrite:
            self.expect_kw("INTO")
        else:
            self.eat_kw("TABLE")
        self.eat_kw("TABLE")
        table = self._qualified_name()
        cols = None
        if self.at_op("(") and not self._peek_is_query_paren() \
                and not (self.peek(1).kind == "ident"
                         and self.peek(1).upper == "VALUES"):
            self.expect_op("(")
            cols = [self.ident()]
            while self.eat_op(","):
                cols.append(self.ident())
            self.expect_op(")")
        return S.InsertInto(table=table, input=self.parse_query(),
                            overwrite=overwrite, columns=cols)

    def _parse_merge(self) -> S.Plan:
        self.expect_kw("MERGE")
        self.expect_kw("INTO")
        target = self._qualified_name()
        talias = None
        if self.eat_kw("AS"):
            talias = self.ident()
        elif self.peek().kind == "ident" and self.peek().upper not in ("USING",):
            talias = self.ident()
        self.expect_kw("USING")
        if self.at_op("("):
            self.expect_op("(")
            source = self.parse_query()
            self.expect_op(")")
        else:
            source = S.Read(table=self._qualified_name())
        salias = None
        if self.eat_kw("AS"):
            salias = self.ident()
        elif self.peek().kind == "ident" and self.peek().upper not in ("ON",):
            salias = self.ident()
        self.expect_kw("ON")
        on = self.parse_expr()
        matched, not_matched, nm_by_source = [], [], []
        while self.at_kw("WHEN"):
            self.next()
            if self.eat_kw("MATCHED"):
                cond = self.parse_expr() if self.eat_kw("AND") else None
                self.expect_kw("THEN")
                if self.eat_kw("DELETE"):
                    matched.append(S.MergeAction("delete", cond))
                else:
                    self.expect_kw("UPDATE")
                    self.expect_kw("SET")
                    if self.at_op("*"):
                        self.next()
                        matched.append(S.MergeAction("update_star", cond))
                    else:
                        matched.append(S.MergeAction("update", cond,
                                                     self._parse_assignments()))
            else:
                self.expect_kw("NOT")
                self.expect_kw("MATCHED")
                by_source = False
                if self.eat_kw("BY"):
                    if self.eat_kw("SOURCE"):
                        by_source = True
                    else:
                        self.expect_kw("TARGET")
                cond = self.parse_expr() if self.eat_kw("AND") else None
                self.expect_kw("THEN")
                if by_source:
                    if self.eat_kw("DELETE"):
                        nm_by_source.append(S.MergeAction("delete", cond))
                    else:
                        self.expect_kw("UPDATE")
                        self.expect_kw("SET")
                        nm_by_source.append(S.MergeAction("update", cond,
                                                          self._parse_assignments()))
                else:
                    self.expect_kw("INSERT")
                    if self.at_op("*"):
                        self.next()
                        not_matched.append(S.MergeAction("insert_star", cond))
                    else:
                        self.expect_op("(")
                        cols = [self.ident()]
                        while self.eat_op(","):
                            cols.append(self.ident())
                        self.expect_op(")")
                        self.expect_kw("VALUES")
                        self.expect_op("(")
                        vals = self._expr_list()
                        self.expect_op(")")
                        not_matched.append(S.MergeAction("insert", cond,
                                                         insert_columns=cols,
                                                         insert_values=vals))
        return S.MergeInto(target=target, target_alias=talias, source=source,
                           source_alias=salias, on=on, matched=matched,
                           not_matched=not_matched, not_matched_by_source=nm_by_source)

    def _parse_assignments(self):
        out = []
        while True:
            name = self._qualified_name()
            self.expect_op("=")
            out.append((name, self.parse_expr()))
            if not self.eat_op(","):
                break
        return out

    def _parse_update(self) -> S.Plan:
        self.expect_kw("UPDATE")
        table = self._qualified_name()
        self.expect_kw("SET")
        assignments = self._parse_assignments()
        cond = self.parse_expr() if self.eat_kw("WHERE") else None
        return S.UpdateTable(table=table, assignments=assignments, condition=cond)

    def _parse_delete(self) -> S.Plan:
        self.expect_kw("DELETE")
        self.expect_kw("FROM")
        table = self._qualified_name()
        cond = self.parse_expr() if self.eat_kw("WHERE") else None
        return S.DeleteFrom(table=table, condition=cond)

    def _qualified_name(self) -> str:
        parts = [self.ident()]
        while self.eat_op("."):
            parts.append(self.ident())
        return ".".join(parts)

    # =======================================================================
    # Queries
    # =======================================================================
    def parse_query(self) -> S.Plan:
        ctes: List[Tuple[str, S.Plan]] = []
        if self.eat_kw("WITH"):
            recursive = self.eat_kw("RECURSIVE")
            while True:
                name = self.ident()
                col_aliases = None
                if self.at_op("("):
                    self.expect_op("(")
                    col_aliases = [self.ident()]
                    while self.eat_op(","):
                        col_aliases.append(self.ident())
                    self.expect_op(")")
                self.expect_kw("AS")
                self.expect_op("(")
                sub = self.parse_query()
                self.expect_op(")")
                if col_aliases:
                    sub = S.SubqueryAlias(input=sub, alias=name, column_aliases=col_aliases)
                ctes.append((name, sub))
                if not self.eat_op(","):
                    break
            body = self._parse_set_query()
            return S.WithCte(ctes=ctes, input=body, recursive=recursive)
        return self._parse_set_query()

    def _parse_set_query(self) -> S.Plan:
        left = self._parse_query_term()
        while True:
            if self.at_kw("UNION", "INTERSECT", "EXCEPT", "MINUS"):
                op = self.next().upper
                is_all = self.eat_kw("ALL")
                if not is_all:
                    self.eat_kw("DISTINCT")
                right = self._parse_query_term()
                opname = {"UNION": "union", "INTERSECT": "intersect", "EXCEPT": "except", "MINUS": "except"}[op]
                left = S.SetOp(op=opname, left=left, right=right, is_all=is_all)
            else:
                break
        # trailing ORDER BY / LIMIT applies to the set-op result
        left = self._parse_order_limit(left)
        return left

    def _parse_query_term(self) -> S.Plan:
        if self.at_op("("):
            self.expect_op("(")
            q = self.parse_query()
            self.expect_op(")")
            return q
        if self.at_kw("VALUES"):
            return self._parse_values()
        if self.at_kw("TABLE"):
            self.next()
            return S.Read(table=self._qualified_name())
        return self._parse_select()

    def _parse_values(self) -> S.Plan:
        self.expect_kw("VALUES")
        rows = []
        while True:
            self.expect_op("(")
            row = [self.parse_expr()]
            while self.eat_op(","):
                row.append(self.parse_expr())
            self.expect_op(")")
            rows.append(row)
            if not self.eat_op(","):
                break
        ncols = len(rows[0])
        try:
            data = {f"col{j+1}": [self._literal_value(rows[r][j]) for r in range(len(rows))]
                    for j in range(ncols)}
            return S.LocalRelation(data=data)
        except SqlError:
            # non-literal row values: UNION ALL of one-row SELECTs, which
            # reuses the full expression resolution/eval machinery
            plan = None
            for row in rows:
                sel = S.Project(input=S.LocalRelation(data={"__one__": [1]}),
                                exprs=[S.Alias(e, f"col{j+1}", None)
                                       for j, e in enumerate(row)])
                plan = sel if plan is None else S.SetOp(op="union", left=plan,
                                                        right=sel, is_all=True)
            return plan

    def _literal_value(self, e: S.Expr):
        if isinstance(e, S.Literal):
            return e.value
        if isinstance(e, S.UnaryOp) and e.op == "neg" and isinstance(e.child, S.Literal):
            return -e.child.value
        if isinstance(e, S.Cast) and isinstance(e.child, S.Literal):
            return e.child.value
        raise SqlError("VALUES rows must be literals (expressions TODO)")

    def _parse_select(self) -> S.Plan:
        self.expect_kw("SELECT")
        distinct = False
        if self.eat_kw("DISTINCT"):
            distinct = True
        else:
            self.eat_kw("ALL")
        projections: List[S.Expr] = []
        while True:
            projections.append(self._parse_projection())
            if not self.eat_op(","):
                break

        plan: Optional[S.Plan] = None
        if self.eat_kw("FROM"):
            plan = self._parse_from()
        else:
            plan = S.LocalRelation(data={"__one__": [1]})  # SELECT without FROM

        if self.eat_kw("WHERE"):
            plan = S.Filter(input=plan, condition=self.parse_expr())

        group_by: List[S.Expr] = []
        grouping_sets = None
        if self.eat_kw("GROUP"):
            self.expect_kw("BY")
            if self.eat_kw("ROLLUP"):
                self.expect_op("(")
                group_by = self._expr_list()
                self.expect_op(")")
                grouping_sets = [list(range(k)) for k in range(len(group_by), -1, -1)]
            elif self.eat_kw("CUBE"):
                self.expect_op("(")
                group_by = self._expr_list()
                self.expect_op(")")
                n_ = len(group_by)
                grouping_sets = [[i for i in range(n_) if m & (1 << i)]
                                 for m in range((1 << n_) - 1, -1, -1)]
            elif self.eat_kw("GROUPING"):
                self.expect_kw("SETS")
                self.expect_op("(")
                sets_exprs = []
                while True:
                    self.expect_op("(")
                    one = [] if self.at_op(")") else self._expr_list()
                    self.expect_op(")")
                    sets_exprs.append(one)
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
                # group_by = union of all expressions, sets = index lists
                keymap = {}
                for se in sets_exprs:
                    for e in se:
                        k = repr(e)
                        if k not in keymap:
                            keymap[k] = (len(group_by), e)
                            group_by.append(e)
                grouping_sets = [[keymap[repr(e)][0] for e in se] for se in sets_exprs]
            elif self.at_kw("ALL") and self.peek(1).upper not in ("(",) \
                    and not (self.peek(1).kind == "op" and self.peek(1).value in (",", "(")):
                self.next()
                # GROUP BY ALL: every non-aggregate select item is a key
                for pe in projections:
                    inner = pe.child if isinstance(pe, S.Alias) else pe
                    if not self._contains_agg(pe) and not isinstance(inner, S.Star):
                        group_by.append(inner)
            else:
                group_by = self._expr_list()
                if self.eat_kw("WITH"):
                    kw = self.next().upper
                    if kw == "ROLLUP":
                        grouping_sets = [list(range(k)) for k in range(len(group_by), -1, -1)]
                    elif kw == "CUBE":
                        n_ = len(group_by)
                        grouping_sets = [[i for i in range(n_) if m & (1 << i)]
                                         for m in range((1 << n_) - 1, -1, -1)]

        having = None
        if self.eat_kw("HAVING"):
            having = self.parse_expr()

        if self.at_kw("WINDOW"):
            # named windows: WINDOW w AS (spec) [, w2 AS (spec)]
            self.next()
            named = {}
            while True:
                wname = self.ident()
                self.expect_kw("AS")
                part, order, frame = self._parse_window_spec()
                named[wname.lower()] = (part, order, frame)
                if not self.eat_op(","):
                    break

            def patch(e):
                ref = e.__dict__.get("_window_ref") if hasattr(e, "__dict__") else None
                if isinstance(e, S.WindowExpr) and ref:
                    spec = named.get(ref.lower())
                    if spec is None:
                        raise SqlError(f"undefined window {ref}", self.sql, 0)
                    e.partition_by, e.order_by, e.frame = spec
                for c in e.children():
                    patch(c)

            for pe in projections:
                patch(pe)

        # Build: aggregate if group_by or aggregate functions present
        has_agg = group_by or any(self._contains_agg(p) for p in projections) or (
            having is not None and self._contains_agg(having))
        if has_agg:
            plan = S.Aggregate(input=plan, group_by=group_by, aggs=projections,
                               grouping_sets=grouping_sets, having=having)
        else:
            if having is not None:
                plan = S.Filter(input=plan, condition=having)
            plan = S.Project(input=plan, exprs=projections)
        if distinct:
            plan = S.Distinct(input=plan)
        # ORDER BY / LIMIT are consumed by _parse_set_query so that they bind
        # to the whole set-operation result, not the last SELECT term.
        return plan

    def _parse_order_limit(self, plan: S.Plan) -> S.Plan:
        if self.eat_kw("ORDER"):
            self.expect_kw("BY")
            if self.at_kw("ALL") and self.peek(1).kind != "op" or \
                    (self.at_kw("ALL") and self.peek(1).kind == "op"
                     and self.peek(1).value != "("):
                self.next()
                asc = not self.eat_kw("DESC")
                self.eat_kw("ASC")
                keys = [S.SortKey(S.Col("__all__"), asc, None)]
            else:
                keys = [self._parse_sort_key()]
                while self.eat_op(","):
                    keys.append(self._parse_sort_key())
            plan = S.Sort(input=plan, keys=keys)
        if self.at_kw("CLUSTER"):
            # CLUSTER BY = DISTRIBUTE BY + SORT BY; a whole-partition-per-GPU
            # engine treats the distribution as a physical no-op and keeps
            # the sort
            self.next()
            self.expect_kw("BY")
            keys = [S.SortKey(e, True, None) for e in self._expr_list()]
            plan = S.Sort(input=plan, keys=keys)
        if self.at_kw("DISTRIBUTE"):
            self.next()
            self.expect_kw("BY")
            self._expr_list()  # partitioning hint: no-op locally
        if self.at_kw("SORT"):
            # SORT BY: partition-local sort; equals a total sort here
            self.next()
            self.expect_kw("BY")
            keys = [self._parse_sort_key()]
            while self.eat_op(","):
                keys.append(self._parse_sort_key())
            plan = S.Sort(input=plan, keys=keys)
        if self.eat_kw("LIMIT"):
            if self.eat_kw("ALL"):
                pass
            else:
                n = self.parse_expr()
                if not isinstance(n, S.Literal):
                    raise SqlError("LIMIT must be a literal")
                offset = 0
                if self.eat_kw("OFFSET"):
                    off = self.parse_expr()
                    offset = int(off.value)
                plan = S.Limit(input=plan, n=int(n.value), offset=offset)
        elif self.eat_kw("OFFSET"):
            off = self.parse_expr()
            plan = S.Limit(input=plan, n=None, offset=int(off.value))
        return plan

    def _parse_sort_key(self) -> S.SortKey:
        e = self.parse_expr()
        asc = True
        if self.eat_kw("ASC"):
            asc = True
        elif self.eat_kw("DESC"):
            asc = False
        nulls_first = None
        if self.eat_kw("NULLS"):
            if self.eat_kw("FIRST"):
                nulls_first = True
            else:
                self.expect_kw("LAST")
                nulls_first = False
        return S.SortKey(e, asc, nulls_first)

    def _expr_list(self) -> List[S.Expr]:
        out = [self.parse_expr()]
        while self.eat_op(","):
            out.append(self.parse_expr())
        return out

    def _parse_projection(self) -> S.Expr:
        if self.at_op("*"):
            self.next()
            return S.Star()
        # qualified star: t.*
        if (self.peek().kind == "ident" and self.peek(1).kind == "op" and self.peek(1).value == "."
                and self.peek(2).kind == "op" and self.peek(2).value == "*"):
            q = self.ident()
            self.next()
            self.next()
            return S.Star(qualifier=q)
        e = self.parse_expr()
        if self.eat_kw("AS"):
            return S.Alias(e, self.ident())
        # bare alias: expr ident  (but not reserved words)
        t = self.peek()
        if t.kind == "ident" and t.upper not in _RESERVED_STOP:
            self.next()
            return S.Alias(e, t.value)
        return e

    def _contains_agg(self, e: S.Expr) -> bool:
        if isinstance(e, S.AggFunc):
            return True
        if isinstance(e, S.WindowExpr):
            return False  # window fns are not group aggs
        if isinstance(e, (S.ScalarSubquery, S.InSubquery, S.Exists)):
            return False
        return any(self._contains_agg(c) for c in e.children())

    # -- FROM clause -------------------------------------------------------
    def _parse_from(self) -> S.Plan:
        plan = self._parse_table_factor()
        while True:
            if self.at_kw("LATERAL") and self.peek(1).upper == "VIEW":
                self.next()
                self.next()
                outer = self.eat_kw("OUTER")
                fn_name = self.ident().lower()
                self.expect_op("(")
                args = [] if self.at_op(")") else self._expr_list()
                self.expect_op(")")
                view_alias = self.ident() if self.peek().kind == "ident" \
                    and self.peek().upper != "AS" else None
                aliases = None
                if self.eat_kw("AS"):
                    aliases = [self.ident()]
                    while self.eat_op(","):
                        aliases.append(self.ident())
                plan = S.Generate(
                    input=plan, gen=S.Func(fn_name, args),
                    outer=outer or fn_name.endswith("_outer"),
                    position=fn_name.startswith("posexplode"),
                    aliases=aliases, view_alias=view_alias)
                continue
            if self.at_kw("PIVOT"):
                self.next()
                self.expect_op("(")
                agg = self.parse_expr()
                self.expect_kw("FOR")
                pivot = self._parse_additive()  # stop before IN
                self.expect_kw("IN")
                self.expect_op("(")
                values = []
                while True:
                    v = self.parse_expr()
                    if self.eat_kw("AS"):
                        v = S.Alias(v, self.ident())
                    elif self.peek().kind == "ident" and self.peek().upper not in _RESERVED_STOP:
                        v = S.Alias(v, self.ident())
                    values.append(v)
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
                self.expect_op(")")
                plan = S.Pivot(input=plan, agg=agg, pivot=pivot, values=values)
                continue
            if self.at_kw("UNPIVOT"):
                self.next()
                self.expect_op("(")
                value_name = self.ident()
                self.expect_kw("FOR")
                name_name = self.ident()
                self.expect_kw("IN")
                self.expect_op("(")
                cols = [self.ident()]
                while self.eat_op(","):
                    cols.append(self.ident())
                self.expect_op(")")
                self.expect_op(")")
                plan = S.Unpivot(input=plan, value_name=value_name,
                                 name_name=name_name, columns=cols)
                continue
            if self.eat_op(","):
                right = self._parse_table_factor()
                plan = S.Join(left=plan, right=right, how="cross", on=None)
                continue
            natural = self.eat_kw("NATURAL")
            jt = None
            if self.at_kw("JOIN"):
                jt = "inner"
            elif self.at_kw("INNER"):
                self.next()
                jt = "inner"
            elif self.at_kw("LEFT"):
                self.next()
                self.eat_kw("OUTER")
                jt = "left"
                if self.eat_kw("SEMI"):
                    jt = "semi"
                elif self.eat_kw("ANTI"):
                    jt = "anti"
            elif self.at_kw("RIGHT"):
                self.next()
                self.eat_kw("OUTER")
                jt = "right"
                if self.eat_kw("SEMI"):
                    jt = "rightsemi"
                elif self.eat_kw("ANTI"):
                    jt = "rightanti"
            elif self.at_kw("FULL"):
                self.next()
                self.eat_kw("OUTER")
                jt = "full"
            elif self.at_kw("CROSS"):
                self.next()
                jt = "cross"
            elif self.at_kw("SEMI"):
                self.next()
                jt = "semi"
            elif self.at_kw("ANTI"):
                self.next()
                jt = "anti"
            if jt is None:
                break
            self.expect_kw("JOIN")
            right = self._parse_table_factor()
            on = None
            using = None
            if jt != "cross" and not natural:
                if self.eat_kw("ON"):
                    on = self.parse_expr()
                elif self.eat_kw("USING"):
                    self.expect_op("(")
                    using = [self.ident()]
                    while self.eat_op(","):
                        using.append(self.ident())
                    self.expect_op(")")
            plan = S.Join(left=plan, right=right, how=jt, on=on, using=using or (["__natural__"] if natural else None))
        return plan

    def _parse_table_factor(self) -> S.Plan:
        lateral = self.eat_kw("LATERAL")
        if self.at_op("("):
            self.expect_op("(")
            sub = self.parse_query()
            self.expect_op(")")
            plan = self._maybe_sample(sub)
            alias, cols = self._parse_alias()
            if alias:
                out = S.SubqueryAlias(input=plan, alias=alias,
                                      column_aliases=cols)
            else:
                out = plan
            if lateral:
                out.__dict__["_lateral"] = True
            return out
        if self.at_kw("VALUES"):
            plan = self._parse_values()
            alias, cols = self._parse_alias()
            if alias:
                return S.SubqueryAlias(input=plan, alias=alias, column_aliases=cols)
            return plan
        name = self._qualified_name()
        if self.at_op("("):
            # table function: range(...), explode(...) etc.
            self.expect_op("(")
            args = [] if self.at_op(")") else self._expr_list()
            self.expect_op(")")
            plan = self._maybe_sample(self._table_function(name, args))
            alias, cols = self._parse_alias()
            if alias:
                return S.SubqueryAlias(input=plan, alias=alias, column_aliases=cols)
            return plan
        rd = S.Read(table=name)
        tt = self._maybe_temporal()
        if tt:
            rd.options.update(tt)
        plan = self._maybe_sample(rd)
        alias, cols = self._parse_alias()
        if alias:
            return S.SubqueryAlias(input=plan, alias=alias, column_aliases=cols)
        return plan

    def _maybe_temporal(self):
        """[FOR] (VERSION | SYSTEM_VERSION) AS OF v  |
        [FOR] (TIMESTAMP | SYSTEM_TIME) AS OF ts  -> lakehouse time-travel
        options (ref: sail-sql-parser ast/query.rs TemporalClause;
        delta/iceberg readers honour versionAsOf/timestampAsOf)."""
        t = self.peek()
        off = 0
        if t.kind == "ident" and t.upper == "FOR":
            t = self.peek(1)
            off = 1
        if t.kind != "ident" or t.upper not in (
                "VERSION", "SYSTEM_VERSION", "TIMESTAMP", "SYSTEM_TIME"):
            return None
        if not (self.peek(off + 1).kind == "ident"
                and self.peek(off + 1).upper == "AS"
                and self.peek(off + 2).kind == "ident"
                and self.peek(off + 2).upper == "OF"):
            return None
        for _ in range(off + 3):
            self.next()
        e = self.parse_expr()
        if not isinstance(e, S.Literal):
            raise SqlError("AS OF expects a literal version/timestamp")
        v = e.value
        if t.upper in ("VERSION", "SYSTEM_VERSION"):
            return {"versionAsOf": str(int(v))}
        if isinstance(e.dtype, T.TimestampType):
            return {"timestampAsOf": v / 1_000_000}  # micros -> seconds
        return {"timestampAsOf": v}

    def _maybe_sample(self, plan: S.Plan) -> S.Plan:
        """TABLESAMPLE (n PERCENT | n ROWS) [REPEATABLE (seed)]"""
        if not self.eat_kw("TABLESAMPLE"):
            return plan
        self.expect_op("(")
        t = self.next()
        if t.kind != "number":
            raise SqlError("TABLESAMPLE expects a number")
        val = float(t.value.split("#")[0])
        fraction = rows = None
        if self.eat_kw("PERCENT"):
            fraction = val / 100.0
        elif self.eat_kw("ROWS"):
            rows = int(val)
        else:
            raise SqlError("TABLESAMPLE expects PERCENT or ROWS")
        self.expect_op(")")
        seed = None
        if self.eat_kw("REPEATABLE"):
            self.expect_op("(")
            st = self.next()
            seed = int(st.value.split("#")[0])
            self.expect_op(")")
        return S.Sample(input=plan, fraction=fraction, rows=rows, seed=seed)

    def _table_function(self, name: str, args: List[S.Expr]) -> S.Plan:
        lname = name.lower()
        if lname == "range":
            vals = [int(a.value) for a in args if isinstance(a, S.Literal)]
            if len(vals) == 1:
                return S.Range(0, vals[0], 1)
            if len(vals) == 2:
                return S.Range(vals[0], vals[1], 1)
            return S.Range(vals[0], vals[1], vals[2])
        return S.TableFuncRead(name=lname, args=args)

    def _parse_alias(self):
        if self.eat_kw("AS"):
            name = self.ident()
        else:
            t = self.peek()
            if t.kind == "ident" and t.upper not in _RESERVED_STOP and t.upper not in _JOIN_TYPES:
                name = self.ident()
            else:
                return None, None
        cols = None
        if self.at_op("("):
            self.expect_op("(")
            cols = [self.ident()]
            while self.eat_op(","):
                cols.append(self.ident())
            self.expect_op(")")
        return name, cols

    # =======================================================================
    # Expressions (Pratt)
    # =======================================================================
    def parse_expr(self) -> S.Expr:
        # lambda: `x -> expr` or `(x, y) -> expr`
        t = self.peek()
        if t.kind == "ident" and self.peek(1).kind == "op" and self.peek(1).value == "->":
            name = self.ident()
            self.next()  # ->
            return S.Lambda([name], self.parse_expr())
        if self.at_op("("):
            save = self.i
            if self._try_lambda_params() is not None:
                params = self._try_lambda_params(consume=True)
                return S.Lambda(params, self.parse_expr())
            self.i = save
        return self._parse_or()

    def _try_lambda_params(self, consume: bool = False):
        """Look ahead for `(a, b, ...) ->`; returns params or None."""
        save = self.i
        if not self.eat_op("("):
            self.i = save
            return None
        params = []
        while True:
            t = self.peek()
            if t.kind != "ident":
                self.i = save
                return None
            params.append(t.value)
            self.next()
            if self.eat_op(","):
                continue
            break
        if not self.eat_op(")") or not (self.at_op("->")):
            self.i = save
            return None
        self.next()  # ->
        if not consume:
            self.i = save
        return params

    def _parse_or(self) -> S.Expr:
        left = self._parse_and()
        while self.eat_kw("OR"):
            left = S.BinaryOp("or", left, self._parse_and())
        return left

    def _parse_and(self) -> S.Expr:
        left = self._parse_not()
        while self.eat_kw("AND"):
            left = S.BinaryOp("and", left, self._parse_not())
        return left

    def _parse_not(self) -> S.Expr:
        if self.eat_kw("NOT") or self.eat_op("!"):
            return S.UnaryOp("not", self._parse_not())
        return self._parse_predicate()

    def _parse_predicate(self) -> S.Expr:
        if self.at_kw("EXISTS") and self.peek(1).kind == "op" and self.peek(1).value == "(" \
                and self.peek(2).kind == "ident" and self.peek(2).upper in ("SELECT", "WITH"):
            self.next()
            self.expect_op("(")
            sub = self.parse_query()
            self.expect_op(")")
            return S.Exists(plan=sub)
        left = self._parse_comparison()
        while True:
            negated = False
            save = self.i
            if self.eat_kw("NOT"):
                negated = True
            if self.eat_kw("BETWEEN"):
                lo = self._parse_additive()
                self.expect_kw("AND")
                hi = self._parse_additive()
                left = S.Between(left, lo, hi, negated)
                continue
            if self.at_kw("LIKE", "ILIKE", "RLIKE", "REGEXP"):
                kw = self.next().upper
                pat = self._parse_additive()
                left = S.Like(left, pat, negated, case_insensitive=(kw == "ILIKE"),
                              is_regex=(kw in ("RLIKE", "REGEXP")))
                continue
            if self.eat_kw("IN"):
                self.expect_op("(")
                if self.at_kw("SELECT", "WITH", "VALUES"):
                    sub = self.parse_query()
                    self.expect_op(")")
                    left = S.InSubquery(left, sub, negated)
                else:
                    vals = self._expr_list()
                    self.expect_op(")")
                    left = S.InList(left, vals, negated)
                continue
            if negated:
                self.i = save
            break
        # IS [NOT] NULL / TRUE / FALSE / DISTINCT FROM
        while self.at_kw("IS"):
            self.next()
            neg = self.eat_kw("NOT")
            if self.eat_kw("NULL"):
                left = S.UnaryOp("isnotnull" if neg else "isnull", left)
            elif self.eat_kw("TRUE"):
                e = S.BinaryOp("<=>", left, S.Literal(True, T.BOOL))
                left = S.UnaryOp("not", e) if neg else e
            elif self.eat_kw("FALSE"):
                e = S.BinaryOp("<=>", left, S.Literal(False, T.BOOL))
                left = S.UnaryOp("not", e) if neg else e
            elif self.eat_kw("DISTINCT"):
                self.expect_kw("FROM")
                rhs = self._parse_comparison()
                e = S.UnaryOp("not", S.BinaryOp("<=>", left, rhs))
                left = S.UnaryOp("not", e) if neg else e
            else:
                raise SqlError("expected NULL/TRUE/FALSE/DISTINCT after IS", self.sql, self.peek().pos)
        return left

    def _parse_comparison(self) -> S.Expr:
        left = self._parse_additive()
        while self.at_op("=", "!=", "<>", "<", "<=", ">", ">=", "<=>"):
            op = self.next().value
            if op == "<>":
                op = "!="
            if self.at_kw("ANY", "SOME", "ALL") and self.peek(1).value == "(":
                left = self._quantified_cmp(op, left)
                continue
            right = self._parse_additive()
            left = S.BinaryOp(op, left, right)
        return left

    def _quantified_cmp(self, op: str, left: S.Expr) -> S.Expr:
        """x op ANY/SOME/ALL (subquery) -> EXISTS rewrite over a one-column
        subquery (ref: Spark quantified predicates). Known deviation: a
        NULL in the subquery column yields FALSE where Spark yields NULL."""
        q = self.next().upper
        self.expect_op("(")
        sub = self.parse_query()
        self.expect_op(")")
        aliased = S.SubqueryAlias(input=sub, alias="__qnt",
                                  column_aliases=["__qc"])
        col = S.Col("__qc", qualifier="__qnt")
        if q in ("ANY", "SOME"):
            cond = S.BinaryOp(op, left, col)
            return S.Exists(plan=S.Filter(input=aliased, condition=cond))
        # ALL: no counterexample rows (violations OR null comparisons)
        cmp_ = S.BinaryOp(op, left, col)
        bad = S.BinaryOp("or", S.UnaryOp("not", cmp_),
                         S.UnaryOp("isnull", cmp_))
        return S.Exists(plan=S.Filter(input=aliased, condition=bad),
                        negated=True)

    def _parse_additive(self) -> S.Expr:
        left = self._parse_multiplicative()
        while True:
            if self.at_op("+", "-"):
                op = self.next().value
                left = S.BinaryOp(op, left, self._parse_multiplicative())
            elif self.at_op("||"):
                self.next()
                left = S.Func("concat", [left, self._parse_multiplicative()])
            else:
                break
        return left

    def _parse_multiplicative(self) -> S.Expr:
        left = self._parse_unary()
        while self.at_op("*", "/", "%") or self.at_kw("DIV"):
            if self.at_kw("DIV"):
                self.next()
                left = S.BinaryOp("div", left, self._parse_unary())
            else:
                op = self.next().value
                left = S.BinaryOp(op, left, self._parse_unary())
        return left

    def _parse_unary(self) -> S.Expr:
        if self.at_op("-"):
            self.next()
            return S.UnaryOp("neg", self._parse_unary())
        if self.at_op("+"):
            self.next()
            return self._parse_unary()
        if self.at_op("~"):
            self.next()
            return S.Func("bitwise_not", [self._parse_unary()])
        return self._parse_postfix()

    def _parse_postfix(self) -> S.Expr:
        e = self._parse_primary()
        while True:
            if self.eat_op("::"):
                e = S.Cast(e, self._parse_type())
            elif self.at_op("[") :
                self.next()
                idx = self.parse_expr()
                self.expect_op("]")
                e = S.Func("element_at_sql", [e, idx])
            elif self.at_op(".") and self.peek(1).kind == "ident":
                # struct field access a.b (only when `a` is not a plain column
                # ref — qualified columns are handled in _parse_primary)
                self.next()
                e = S.Func("get_field", [e, S.Literal(self.ident(), T.STRING)])
            else:
                break
        return e

    # -- primary -----------------------------------------------------------
    def _parse_primary(self) -> S.Expr:
        t = self.peek()
        if t.kind == "number":
            self.next()
            return _number_literal(t.value)
        if t.kind == "string":
            self.next()
            return S.Literal(t.value, T.STRING)
        if self.at_op("("):
            self.next()
            if self.at_kw("SELECT", "WITH"):
                sub = self.parse_query()
                self.expect_op(")")
                return S.ScalarSubquery(plan=sub)
            e = self.parse_expr()
            if self.at_op(","):
                # row constructor (a, b, ...) — used in IN ((1,2),(3,4)); keep as struct
                items = [e]
                while self.eat_op(","):
                    items.append(self.parse_expr())
                self.expect_op(")")
                return S.Func("struct", items)
            self.expect_op(")")
            return e
        if self.at_op("*"):
            self.next()
            return S.Star()
        if self.at_op("?"):
            self.next()
            return S.Literal(None, T.NULL)
        if t.kind != "ident":
            raise SqlError(f"unexpected token {t.value!r}", self.sql, t.pos)

        kw = t.upper
        if kw == "CASE":
            return self._parse_case()
        if kw == "CAST" or kw == "TRY_CAST":
            self.next()
            self.expect_op("(")
            e = self.parse_expr()
            self.expect_kw("AS")
            ty = self._parse_type()
            self.expect_op(")")
            return S.Cast(e, ty, try_=(kw == "TRY_CAST"))
        if kw == "EXTRACT":
            self.next()
            self.expect_op("(")
            fld = self.ident().lower()
            self.expect_kw("FROM")
            e = self.parse_expr()
            self.expect_op(")")
            return S.Func(fld if fld in ("year", "month", "day", "hour", "minute", "second", "quarter", "week") else "date_part_" + fld, [e])
        if kw == "INTERVAL":
            return self._parse_interval()
        if kw == "X" and self.peek(1).kind == "string":
            self.next()
            s = self.next().value.replace(" ", "")
            return S.Literal(bytes.fromhex(s), T.BINARY)
        if kw == "DATE" and self.peek(1).kind == "string":
            self.next()
            s = self.next().value
            return S.Literal(_parse_date(s), T.DATE)
        if kw == "TIMESTAMP" and self.peek(1).kind == "string":
            self.next()
            s = self.next().value
            return S.Literal(_parse_timestamp_us(s), T.TIMESTAMP)
        if kw == "NULL":
            self.next()
            return S.Literal(None, T.NULL)
        if kw == "TRUE":
            self.next()
            return S.Literal(True, T.BOOL)
        if kw == "FALSE":
            self.next()
            return S.Literal(False, T.BOOL)
        if kw == "EXISTS" and self.peek(2).kind == "ident" \
                and self.peek(2).upper in ("SELECT", "WITH"):
            # subquery EXISTS (predicate path covers the normal spot; this
            # one appears after NOT). exists(arr, lambda) falls through to
            # the generic function-call parse.
            self.next()
            self.expect_op("(")
            sub = self.parse_query()
            self.expect_op(")")
            return S.Exists(plan=sub)
        if kw == "SUBSTRING" and self.peek(1).kind == "op" and self.peek(1).value == "(":
            self.next()
            self.expect_op("(")
            e = self.parse_expr()
            if self.eat_kw("FROM"):
                start = self.parse_expr()
                length = None
                if self.eat_kw("FOR"):
                    length = self.parse_expr()
            else:
                self.expect_op(",")
                start = self.parse_expr()
                length = None
                if self.eat_op(","):
                    length = self.parse_expr()
            self.expect_op(")")
            args = [e, start] + ([length] if length is not None else [])
            return S.Func("substring", args)
        if kw in ("CURRENT_DATE", "CURRENT_TIMESTAMP", "CURRENT_USER") and not (
                self.peek(1).kind == "op" and self.peek(1).value == "("):
            self.next()
            return S.Func(kw.lower(), [])

        # identifier: function call or column
        name = self.ident()
        if self.at_op("(") and (name.upper() not in _RESERVED_STOP
                                or name.upper() == "WINDOW"):
            # WINDOW is reserved for the WINDOW clause but is also the
            # tumbling-window grouping function window(ts, '1 hour')
            return self._parse_call(name)
        # qualified column a.b or a.b.c
        qualifier = None
        while self.at_op(".") and self.peek(1).kind == "ident":
            save = self.i
            self.next()
            nxt = self.ident()
            if self.at_op("(") :
                # db.func(...) — treat last part as function
                return self._parse_call(nxt)
            if qualifier is None:
                qualifier = name
                name = nxt
            else:
                # 3-part: treat as struct field access on qualified col
                return S.Func("get_field", [S.Col(name, qualifier), S.Literal(nxt, T.STRING)])
        return S.Col(name, qualifier)

    def _parse_call(self, name: str) -> S.Expr:
        lname = name.lower()
        self.expect_op("(")
        distinct = False
        if self.eat_kw("DISTINCT"):
            distinct = True
        else:
            self.eat_kw("ALL")
        args: List[S.Expr] = []
        if lname == "position" and not self.at_op(")"):
            # POSITION(needle IN haystack); needle parses below the
            # IN-predicate level so IN stays the separator
            first = self._parse_comparison()
            if self.eat_kw("IN"):
                hay = self.parse_expr()
                self.expect_op(")")
                return S.Func("position", [first, hay])
            args = [first]
            while self.eat_op(","):
                args.append(self.parse_expr())
            self.expect_op(")")
            return S.Func("position", args)
        if lname == "trim" and self.at_kw("BOTH", "LEADING", "TRAILING"):
            # TRIM([BOTH|LEADING|TRAILING] [chars] FROM s)
            side = self.next().upper
            chars = None
            if not self.at_kw("FROM"):
                chars = self.parse_expr()
            self.expect_kw("FROM")
            s_ = self.parse_expr()
            self.expect_op(")")
            fn = {"BOTH": "btrim", "LEADING": "ltrim",
                  "TRAILING": "rtrim"}[side]
            return S.Func(fn, [s_] + ([chars] if chars is not None else []))
        if lname == "overlay" and not self.at_op(")"):
            first = self.parse_expr()
            if self.at_kw("PLACING"):
                # OVERLAY(s PLACING r FROM p [FOR l])
                self.next()
                rep = self.parse_expr()
                self.expect_kw("FROM")
                pos = self.parse_expr()
                a = [first, rep, pos]
                if self.eat_kw("FOR"):
                    a.append(self.parse_expr())
                self.expect_op(")")
                return S.Func("overlay", a)
            args = [first]
            while self.eat_op(","):
                args.append(self.parse_expr())
            self.expect_op(")")
            return S.Func("overlay", args)
        if lname in ("timestampadd", "timestampdiff", "timestamp_add",
                     "timestamp_diff", "date_add_unit") and \
                self.peek().kind == "ident" and \
                self.peek(1).kind == "op" and self.peek(1).value == ",":
            # unit keyword first argument: timestampadd(HOUR, 2, ts)
            args.append(S.Literal(self.next().value.upper(), None))
            self.expect_op(",")
            args += self._expr_list()
            self.expect_op(")")
            return S.Func(lname.replace("timestamp_", "timestamp"), args)
        if not self.at_op(")"):
            if self.at_op("*"):
                self.next()
                args = [S.Star()]
            else:
                args = self._expr_list()
        self.expect_op(")")
        if lname == "equal_null" and len(args) == 2:
            return S.BinaryOp("<=>", args[0], args[1])
        if lname == "nullifzero" and len(args) == 1:
            return S.Func("nullif", [args[0], S.Literal(0, None)])
        if lname == "zeroifnull" and len(args) == 1:
            return S.Func("coalesce", [args[0], S.Literal(0, None)])
        self.eat_kw("IGNORE") and self.expect_kw("NULLS")
        # ordered-set aggregates: f(q) WITHIN GROUP (ORDER BY x [DESC])
        if self.at_kw("WITHIN"):
            self.next()
            self.expect_kw("GROUP")
            self.expect_op("(")
            self.expect_kw("ORDER")
            self.expect_kw("BY")
            order = self.parse_expr()
            desc = False
            if self.eat_kw("DESC"):
                desc = True
            else:
                self.eat_kw("ASC")
            self.expect_op(")")
            if lname in ("percentile_cont", "percentile_disc", "percentile"):
                args = [order] + args + [S.Literal(desc, T.BOOL)]
            elif lname in ("listagg", "string_agg"):
                args = args + [order, S.Literal(desc, T.BOOL)]
            else:
                raise SqlError(f"WITHIN GROUP not supported for {name}",
                               self.sql, self.peek().pos)
        # FILTER (WHERE ...)
        filt = None
        if self.at_kw("FILTER"):
            self.next()
            self.expect_op("(")
            self.expect_kw("WHERE")
            filt = self.parse_expr()
            self.expect_op(")")

        from ..functions.registry import AGG_FUNCTIONS, WINDOW_FUNCTIONS
        from ..engine.aggregates import UDAFS
        e: S.Expr
        if lname in AGG_FUNCTIONS or lname in UDAFS:
            e = S.AggFunc(lname, args, distinct=distinct, filter=filt)
        elif lname in WINDOW_FUNCTIONS:
            e = S.Func(lname, args)
        else:
            e = S.Func(lname, args)

        if self.at_kw("OVER"):
            self.next()
            if self.peek().kind == "ident" and self.peek().upper not in _RESERVED_STOP:
                ref = self.ident()
                e = S.WindowExpr(func=e)
                e.__dict__["_window_ref"] = ref
            else:
                part, order, frame = self._parse_window_spec()
                e = S.WindowExpr(func=e, partition_by=part, order_by=order, frame=frame)
        return e

    def _parse_window_spec(self):
        self.expect_op("(")
        part: List[S.Expr] = []
        order: List[S.SortKey] = []
        frame = None
        if self.eat_kw("PARTITION"):
            self.expect_kw("BY")
            part = self._expr_list()
        if self.eat_kw("ORDER"):
            self.expect_kw("BY")
            order = [self._parse_sort_key()]
            while self.eat_op(","):
                order.append(self._parse_sort_key())
        if self.at_kw("ROWS", "RANGE"):
            mode = self.next().upper.lower()
            lo, hi = self._parse_frame_bounds()
            frame = (mode, lo, hi)
        self.expect_op(")")
        return part, order, frame

    def _parse_frame_bounds(self):
        def bound():
            if self.eat_kw("UNBOUNDED"):
                if self.eat_kw("PRECEDING"):
                    return ("unbounded_preceding", None)
                self.expect_kw("FOLLOWING")
                return ("unbounded_following", None)
            if self.eat_kw("CURRENT"):
                self.expect_kw("ROW")
                return ("current", None)
            n = self.parse_expr()
            v = int(n.value) if isinstance(n, S.Literal) else 0
            if self.eat_kw("PRECEDING"):
                return ("preceding", v)
            self.expect_kw("FOLLOWING")
            return ("following", v)

        if self.eat_kw("BETWEEN"):
            lo = bound()
            self.expect_kw("AND")
            hi = bound()
            return lo, hi
        lo = bound()
        return lo, ("current", None)

    def _parse_case(self) -> S.Expr:
        self.expect_kw("CASE")
        operand = None
        if not self.at_kw("WHEN"):
            operand = self.parse_expr()
        branches = []
        while self.eat_kw("WHEN"):
            cond = self.parse_expr()
            self.expect_kw("THEN")
            val = self.parse_expr()
            if operand is not None:
                cond = S.BinaryOp("=", operand, cond)
            branches.append((cond, val))
        els = None
        if self.eat_kw("ELSE"):
            els = self.parse_expr()
        self.expect_kw("END")
        return S.CaseWhen(branches, els)

    def _parse_interval(self) -> S.Expr:
        """INTERVAL '3' MONTH / INTERVAL 90 DAY / INTERVAL '1-2' YEAR TO MONTH."""
        self.expect_kw("INTERVAL")
        t = self.next()
        if t.kind == "string":
            amount_str = t.value
        elif t.kind == "number":
            amount_str = t.value.split("#")[0]
        elif t.kind == "op" and t.value == "-":
            t2 = self.next()
            amount_str = "-" + (t2.value if t2.kind != "number" else t2.value.split("#")[0])
        else:
            raise SqlError("expected interval amount", self.sql, t.pos)
        unit = self.ident().lower().rstrip("s") if self.peek().kind == "ident" else "day"
        if self.eat_kw("TO"):
            # multi-unit literals: '1-2' YEAR TO MONTH, 'd hh:mm:ss[.f]'
            # DAY TO SECOND (and the HOUR/MINUTE prefixes)
            end_unit = self.ident().lower().rstrip("s")
            return self._multi_unit_interval(amount_str, unit, end_unit, t)
        amount = float(amount_str)
        # Intervals are represented as (months, microseconds) literal pairs;
        # arithmetic resolves them against date/timestamp operands.
        months = 0
        micros = 0
        if unit == "year":
            months = int(amount * 12)
        elif unit == "month":
            months = int(amount)
        elif unit == "week":
            micros = int(amount * 7 * 86400 * 1e6)
        elif unit == "day":
            micros = int(amount * 86400 * 1e6)
        elif unit == "hour":
            micros = int(amount * 3600 * 1e6)
        elif unit == "minute":
            micros = int(amount * 60 * 1e6)
        elif unit == "second":
            micros = int(amount * 1e6)
        else:
            raise SqlError(f"unsupported interval unit {unit}")
        return S.Literal(("__interval__", months, micros), T.NULL)

    def _multi_unit_interval(self, s: str, start: str, end: str, tok):
        neg = s.strip().startswith("-")
        body = s.strip().lstrip("+-")
        months = 0
        micros = 0
        try:
            if start == "year":
                y, _, m = body.partition("-")
                months = int(y) * 12 + (int(m) if m else 0)
            else:
                # [days ]hh:mm:ss[.frac] with start in day/hour/minute
                days = 0
                if start == "day":
                    if " " in body:
                        d, body = body.split(" ", 1)
                        days = int(d)
                    else:
                        days, body = int(body), "0"
                parts = body.split(":")
                secs = 0.0
                mult = {"hour": 3600, "minute": 60, "second": 1}[
                    "hour" if start in ("day", "hour") else start]
                for p in parts:
                    secs += float(p or 0) * mult
                    mult /= 60 if mult > 1 else 1
                    if mult < 1:
                        mult = 1
                micros = int(round((days * 86400 + secs) * 1e6))
        except (ValueError, KeyError):
            raise SqlError(f"bad interval literal {s!r} for "
                           f"{start.upper()} TO {end.upper()}",
                           self.sql, tok.pos)
        if neg:
            months, micros = -months, -micros
        return S.Literal(("__interval__", months, micros), T.NULL)

    def _parse_type(self) -> T.DataType:
        name = self.ident()
        up = name.upper()
        if up in ("ARRAY", "MAP", "STRUCT") and self.at_op("<"):
            self.expect_op("<")
            if up == "ARRAY":
                t = T.ArrayType(self._parse_type())
            elif up == "MAP":
                k = self._parse_type()
                self.expect_op(",")
                t = T.MapType(k, self._parse_type())
            else:
                fields = []
                while True:
                    fn = self.ident()
                    self.eat_op(":")
                    fields.append(T.StructField(fn, self._parse_type()))
                    if not self.eat_op(","):
                        break
                t = T.StructType(tuple(fields))
            # the lexer may tokenize '>>' as one op: split it
            if self.at_op(">>"):
                self.toks[self.i] = Token("op", ">", self.peek().pos)
            else:
                self.expect_op(">")
            return t
        if self.at_op("("):
            self.expect_op("(")
            params = [self.next().value]
            while self.eat_op(","):
                params.append(self.next().value)
            self.expect_op(")")
            return T.type_from_name(f"{name}({','.join(params)})")
        return T.type_from_name(name)

    def _peek_is_query_paren(self) -> bool:
        return self.peek(1).kind == "ident" and self.peek(1).upper in ("SELECT", "WITH")


def _number_literal(value: str) -> S.Expr:
    suffix = ""
    if "#" in value:
        value, suffix = value.split("#")
    if suffix == "BD":
        if "." in value:
            intpart, frac = value.split(".")
            scale = len(frac)
            return S.Literal(float(value), T.DecimalType(max(len(intpart) + scale, scale + 1), scale))
        return S.Literal(float(value), T.DecimalType(len(value), 0))
    if suffix in ("D", "F"):
        return S.Literal(float(value), T.F64 if suffix == "D" else T.F32)
    if "." in value or "e" in value.lower():
        return S.Literal(float(value), T.F64)
    v = int(value)
    if suffix == "L":
        return S.Literal(v, T.I64)
    if suffix == "S":
        return S.Literal(v, T.I16)
    if suffix == "Y":
        return S.Literal(v, T.I8)
    if -(2 ** 31) <= v < 2 ** 31:
        return S.Literal(v, T.I32)
    return S.Literal(v, T.I64)


def _parse_date(s: str) -> int:
    y, m, d = s.strip().split("-")
    return (_dt.date(int(y), int(m), int(d)) - _dt.date(1970, 1, 1)).days


def _parse_timestamp_us(s: str) -> int:
    s = s.strip().replace("T", " ")
    if " " in s:
        datep, timep = s.split(" ", 1)
    else:
        datep, timep = s, "00:00:00"
    y, m, d = (int(x) for x in datep.split("-"))
    frac = 0.0
    parts = timep.split(":")
    hh = int(parts[0]) if parts[0] else 0
    mm = int(parts[1]) if len(parts) > 1 else 0
    ss = float(parts[2]) if len(parts) > 2 else 0.0
    base = _dt.datetime(y, m, d) - _dt.datetime(1970, 1, 1)
    # integer-exact micros: float total_seconds() loses sub-µs precision
    # near the epoch (e.g. 1969-12-31 23:59:59.999999 truncated to 0)
    sec_int = int(ss)
    frac_us = int(round((ss - sec_int) * 1_000_000))
    return ((base.days * 86_400 + hh * 3600 + mm * 60 + sec_int) * 1_000_000
            + frac_us)


def parse_ddl_schema(text: str):
    """'a INT, b STRING' -> [(name, DataType)] (AnalyzePlan DdlParse and
    from_json schema strings)."""
    pr = Parser(text)
    out = []
    while pr.peek().kind != "eof":
        name = pr.ident()
        pr.eat_op(":")
        t = pr._parse_type()
        out.append((name, t))
        if not pr.eat_op(","):
            break
    return out


def parse_sql(sql: str) -> S.Plan:
    """Parse a single SQL statement into the spec IR."""
    stmts = Parser(sql).parse_statements()
    if len(stmts) != 1:
        raise SqlError(f"expected a single statement, got {len(stmts)}")
    return stmts[0]


def parse_sql_multi(sql: str) -> List[S.Plan]:
    return Parser(sql).parse_statements()
