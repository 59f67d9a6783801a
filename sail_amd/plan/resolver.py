"""Plan resolver: spec IR -> typed, bound logical plan.

The analogue of the reference's PlanResolver
(ref: crates/sail-plan/src/resolver/mod.rs:20, resolver/plan.rs:18): binds
column names to input ordinals, infers types, inserts casts, normalizes
aggregates (projection split), inlines CTEs, expands USING/NATURAL joins and
`*`, and marks correlated references as OuterRef for the decorrelator.

The resolved tree uses the same spec node classes with `schema`/`dtype`
filled in and all Col nodes replaced by BoundRef/OuterRef.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field, replace
from typing import Dict, List, Optional, Tuple

from ..engine import types as T
from ..functions.registry import AGG_FUNCTIONS, WINDOW_FUNCTIONS, agg_return_type, scalar_return_type
from . import spec as S


from ..errors import AnalysisException


class ResolutionError(AnalysisException):
    pass


@dataclass
class Field:
    name: str
    dtype: T.DataType
    qualifier: Optional[str] = None


@dataclass
class Scope:
    fields: List[Field] = field(default_factory=list)
    outer: Optional["Scope"] = None

    def find(self, name: str, qualifier: Optional[str]) -> List[int]:
        lname = name.lower()
        lq = qualifier.lower() if qualifier else None
        hits = []
        for i, f in enumerate(self.fields):
            if f.name.lower() != lname:
                continue
            if lq is not None and (f.qualifier or "").lower() != lq:
                continue
            hits.append(i)
        return hits


class Resolver:
    """Resolves a spec plan against a catalog (dict of table -> schema)."""

    def __init__(self, catalog):
        self.catalog = catalog  # engine/session.Catalog
        self.cte_scope: List[Dict[str, S.Plan]] = []
        #: name -> schema while resolving a recursive CTE body
        self.recursion_refs: Dict[str, list] = {}

    # =====================================================================
    def resolve(self, plan: S.Plan) -> S.Plan:
        return self._plan(plan, outer=None)

    # -- plans -------------------------------------------------------------
    def _plan(self, p: S.Plan, outer: Optional[Scope]) -> S.Plan:
        m = getattr(self, "_p_" + type(p).__name__, None)
        if m is None:
            raise ResolutionError(f"cannot resolve plan node {type(p).__name__}")
        return m(p, outer)

    def _scope(self, p: S.Plan, qualifier: Optional[str] = None, outer=None) -> Scope:
        return Scope([Field(n, t, qualifier) for n, t in p.schema], outer)

    def _p_ChunkSource(self, p: S.ChunkSource, outer):
        # pre-materialized leaf (Connect LocalRelation / tail results):
        # schema is already known
        return p

    def _p_Read(self, p: S.Read, outer):
        # working-set reference inside a recursive CTE body?
        if p.table.lower() in self.recursion_refs:
            ref = S.RecursionRef(name=p.table.lower())
            ref.schema = list(self.recursion_refs[p.table.lower()])
            return self._qualify(ref, p.table)
        # CTE reference?
        for scope in reversed(self.cte_scope):
            if p.table.lower() in scope:
                original = scope[p.table.lower()]
                sub = copy.deepcopy(original)
                resolved = self._plan(sub, None) if sub.schema is None else sub
                # every use of this CTE shares one execution-cache token
                resolved.__dict__["_cte_cache_key"] = id(original)
                aliased = S.SubqueryAlias(input=resolved, alias=p.table)
                aliased.schema = resolved.schema
                return self._qualify(aliased, p.table)
        # `SELECT ... FROM parquet.`/path``-style direct file reads (Spark)
        head, _, rest = p.table.partition(".")
        if head.lower() in ("parquet", "csv", "json", "delta", "iceberg") and rest:
            ds = S.DataSourceRead(format=head.lower(), paths=[rest], options=p.options)
            return self._qualify(self._p_DataSourceRead(ds, outer), head.lower())
        schema = self.catalog.table_schema(p.table)
        if schema is None:
            raise ResolutionError(f"table not found: {p.table}")
        out = S.Read(table=p.table, options=p.options)
        out.schema = list(schema)
        base = p.table.split(".")[-1]
        return self._qualify(out, base)

    def _qualify(self, p: S.Plan, name: str) -> S.Plan:
        """Attach a qualifier by wrapping in SubqueryAlias (resolved)."""
        out = S.SubqueryAlias(input=p, alias=name)
        out.schema = p.schema
        return out

    def _p_DataSourceRead(self, p: S.DataSourceRead, outer):
        from ..datasource.registry import infer_source_schema

        out = S.DataSourceRead(format=p.format, paths=p.paths, options=p.options,
                               user_schema=p.user_schema)
        out.schema = p.user_schema or infer_source_schema(p.format, p.paths, p.options)
        return out

    def _p_LocalRelation(self, p: S.LocalRelation, outer):
        out = S.LocalRelation(data=p.data)
        if p.schema:
            out.schema = p.schema
        else:
            out.schema = [(k, _infer_pytype(v)) for k, v in p.data.items()]
        return out

    def _p_TableFuncRead(self, p: S.TableFuncRead, outer):
        udtf = getattr(self.catalog, "udtf", None)
        info = udtf(p.name) if udtf else None
        if info is None:
            raise ResolutionError(f"unknown table function {p.name}")
        out = S.TableFuncRead(name=p.name,
                              args=[self._expr(a, Scope([], outer)) for a in p.args])
        out.schema = list(info[1])
        return self._qualify(out, p.name)

    def _p_Range(self, p: S.Range, outer):
        out = S.Range(p.start, p.end, p.step)
        out.schema = [("id", T.I64)]
        return out

    def _p_WithCte(self, p: S.WithCte, outer):
        scope: Dict[str, S.Plan] = {}
        self.cte_scope.append(scope)
        try:
            for name, sub in p.ctes:
                if p.recursive and _refs_table(sub, name):
                    resolved = self._resolve_recursive_cte(name, sub, outer)
                else:
                    resolved = self._plan(sub, outer)
                scope[name.lower()] = resolved
            return self._plan(p.input, outer)
        finally:
            self.cte_scope.pop()

    def _resolve_recursive_cte(self, name: str, sub: S.Plan, outer):
        """WITH RECURSIVE name AS (anchor UNION [ALL] recursive)
        (ref: crates/sail-plan/src/resolver/query/recursion.rs role)."""
        wrapper_cols = None
        body = sub
        if isinstance(body, S.SubqueryAlias) and body.column_aliases:
            wrapper_cols, body = body.column_aliases, body.input
        if not (isinstance(body, S.SetOp) and body.op == "union"):
            raise ResolutionError(
                f"recursive CTE {name} must be 'anchor UNION [ALL] recursive'")
        anchor = self._plan(body.left, outer)
        if wrapper_cols:
            if len(wrapper_cols) != len(anchor.schema):
                raise ResolutionError(f"recursive CTE {name}: column count mismatch")
            schema = [(wrapper_cols[i], t) for i, (_, t) in enumerate(anchor.schema)]
        else:
            schema = list(anchor.schema)
        self.recursion_refs[name.lower()] = schema
        try:
            rec = self._plan(body.right, outer)
        finally:
            del self.recursion_refs[name.lower()]
        if len(rec.schema) != len(schema):
            raise ResolutionError(f"recursive CTE {name}: column count mismatch")
        if any(rt != st for (_, rt), (_, st) in zip(rec.schema, schema)):
            exprs = []
            for i, ((rn, rt), (_, st)) in enumerate(zip(rec.schema, schema)):
                ref = S.BoundRef(i, rn, rt)
                exprs.append(S.Cast(ref, st, dtype=st) if rt != st else ref)
            pr = S.Project(input=rec, exprs=exprs)
            pr.schema = [(rn, st) for (rn, _), (_, st) in zip(rec.schema, schema)]
            rec = pr
        max_iter = 100
        sess = getattr(self.catalog, "session", None)
        if sess is not None:
            max_iter = int(sess.conf.get("sail.execution.max_recursion", "100"))
        out = S.RecursiveCte(name=name.lower(), anchor=anchor, recursive=rec,
                             is_all=body.is_all, max_iter=max_iter)
        out.schema = schema
        return out

    def _p_SubqueryAlias(self, p: S.SubqueryAlias, outer):
        child = self._plan(p.input, outer)
        names = p.column_aliases or [n for n, _ in child.schema]
        if len(names) != len(child.schema):
            raise ResolutionError(f"alias {p.alias}: {len(names)} aliases for {len(child.schema)} columns")
        out = S.SubqueryAlias(input=child, alias=p.alias, column_aliases=p.column_aliases)
        out.schema = [(names[i], child.schema[i][1]) for i in range(len(names))]
        return out

    def _p_Filter(self, p: S.Filter, outer):
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)
        cond = self._expr(p.condition, scope)
        cond = _coerce_to_bool(cond)
        out = S.Filter(input=child, condition=cond)
        out.schema = child.schema
        return out

    def _p_Project(self, p: S.Project, outer):
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)
        exprs: List[S.Expr] = []
        for e in p.exprs:
            exprs.extend(self._expand_star(e, scope))
        bound = [self._expr(e, scope) for e in exprs]

        # extract window expressions into a WindowPlan below the projection
        windows: List[S.Expr] = []

        def extract_windows(e: S.Expr) -> S.Expr:
            if isinstance(e, S.WindowExpr):
                idx = len(child.schema) + len(windows)
                windows.append(e)
                return S.BoundRef(idx, f"__w{len(windows)-1}", e.dtype)
            if isinstance(e, S.Alias):
                return S.Alias(extract_windows(e.child), e.name, e.dtype)
            ch = e.children()
            if not ch:
                return e
            out = e.with_children([extract_windows(c) for c in ch])
            out.dtype = e.dtype
            return out

        gen_out = self._extract_generator(child, bound)
        if gen_out is not None:
            return gen_out

        bound2 = [extract_windows(e) for e in bound]
        if windows:
            wp = S.WindowPlan(input=child, window_exprs=windows)
            wp.schema = list(child.schema) + [(f"__w{i}", w.dtype) for i, w in enumerate(windows)]
            out = S.Project(input=wp, exprs=bound2)
            out.schema = [(_expr_name(e, i), e.dtype) for i, e in enumerate(bound2)]
            return out
        out = S.Project(input=child, exprs=bound)
        out.schema = [(_expr_name(e, i), e.dtype) for i, e in enumerate(bound)]
        return out

    _GENERATORS = {"explode", "explode_outer", "posexplode",
                   "posexplode_outer", "inline", "inline_outer", "stack"}

    def _p_Pivot(self, p: S.Pivot, outer):
        """PIVOT -> grouped aggregate with one filtered agg per value
        (ref: crates/sail-plan/src/resolver/query/pivoting.rs role)."""
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)
        pivot = self._expr(p.pivot, scope)
        agg_e = p.agg
        alias_name = None
        if isinstance(agg_e, S.Alias):
            alias_name, agg_e = agg_e.name, agg_e.child
        if isinstance(agg_e, S.Func):
            agg_name, agg_args = agg_e.name.lower(), agg_e.args
        elif isinstance(agg_e, S.AggFunc):
            agg_name, agg_args = agg_e.name.lower(), agg_e.args
        else:
            raise ResolutionError("PIVOT expects an aggregate function")
        bound_args = [self._expr(a, scope) for a in agg_args]
        from .rules.util import expr_refs

        used = set()
        for e in [pivot] + bound_args:
            used |= expr_refs(e)
        group_refs = [S.BoundRef(i, n, t) for i, (n, t) in enumerate(child.schema)
                      if i not in used]
        aggs: List[S.Expr] = []
        names: List[str] = []
        for v in p.values:
            vname = None
            ve = v
            if isinstance(v, S.Alias):
                vname, ve = v.name, v.child
            if not isinstance(ve, S.Literal):
                raise ResolutionError("PIVOT IN list expects literals")
            lit = self._expr(ve, scope)
            cond = S.BinaryOp("=", pivot, lit, T.BOOL)
            out_t = agg_return_type(agg_name, [a.dtype for a in bound_args])
            af = S.AggFunc(agg_name, bound_args, False, out_t, cond)
            aggs.append(af)
            base = vname or str(ve.value)
            names.append(base if alias_name is None else f"{base}_{alias_name}")
        out = S.Aggregate(input=child, group_by=group_refs, aggs=aggs)
        out.schema = [(r.name, r.dtype) for r in group_refs] \
            + list(zip(names, [a.dtype for a in aggs]))
        return out

    def _p_Unpivot(self, p: S.Unpivot, outer):
        """UNPIVOT -> UNION ALL of per-column projections + NOT NULL filter
        (Spark default excludeNulls)."""
        child = self._plan(p.input, outer)
        byname = {n.lower(): (i, t) for i, (n, t) in enumerate(child.schema)}
        for c in p.columns:
            if c.lower() not in byname:
                raise ResolutionError(f"UNPIVOT column not found: {c}")
        vt = None
        for c in p.columns:
            vt = byname[c.lower()][1] if vt is None else T.common_type(vt, byname[c.lower()][1])
        unpivot_set = {byname[c.lower()][0] for c in p.columns}
        others = [(i, n, t) for i, (n, t) in enumerate(child.schema)
                  if i not in unpivot_set]
        parts = []
        for c in p.columns:
            ci, ct = byname[c.lower()]
            exprs = [S.BoundRef(i, n, t) for i, n, t in others]
            exprs.append(S.Alias(S.Literal(c, T.STRING), p.name_name, T.STRING))
            vref: S.Expr = S.BoundRef(ci, c, ct)
            if ct != vt:
                vref = S.Cast(vref, vt, dtype=vt)
            exprs.append(S.Alias(vref, p.value_name, vt))
            pr = S.Project(input=child, exprs=exprs)
            pr.schema = [(n, t) for _, n, t in others] \
                + [(p.name_name, T.STRING), (p.value_name, vt)]
            parts.append(pr)
        plan = parts[0]
        for nxt in parts[1:]:
            so = S.SetOp(op="union", left=plan, right=nxt, is_all=True)
            so.schema = parts[0].schema
            plan = so
        vi = len(plan.schema) - 1
        cond = S.UnaryOp("isnotnull",
                         S.BoundRef(vi, p.value_name, vt), dtype=T.BOOL)
        out = S.Filter(input=plan, condition=cond)
        out.schema = plan.schema
        return out

    def _p_Generate(self, p: S.Generate, outer):
        """LATERAL VIEW [OUTER] explode(e) v AS c1[, c2] (ref: Spark
        LATERAL VIEW grammar; generator resolution in
        crates/sail-plan/src/resolver/query/ lateral handling)."""
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)
        fn = p.gen
        if isinstance(fn, S.Func) and fn.name.lower() in self._GENERATORS:
            arr = self._expr(fn.args[0], scope)
        else:
            arr = self._expr(fn, scope)
        if not isinstance(arr.dtype, T.ArrayType):
            raise ResolutionError("LATERAL VIEW generator expects an array")
        elem_t = arr.dtype.element
        aliases = p.aliases
        if aliases is None:
            aliases = ["pos", "col"] if p.position else ["col"]
        want = 2 if p.position else 1
        if len(aliases) != want:
            raise ResolutionError(
                f"LATERAL VIEW {'posexplode' if p.position else 'explode'} "
                f"expects {want} column alias(es), got {len(aliases)}")
        out = S.Generate(input=child, gen=arr, outer=p.outer,
                         position=p.position, aliases=aliases,
                         view_alias=p.view_alias)
        gen_schema = ([(aliases[0], T.I32)] if p.position else []) \
            + [(aliases[-1], elem_t)]
        out.schema = list(child.schema) + gen_schema
        return out

    def _extract_generator(self, child: S.Plan, bound: List[S.Expr]):
        """SELECT explode(arr) [AS x], other... -> Project over Generate.
        The generated element (and pos) columns are appended to the child
        schema, so sibling select items keep their BoundRef ordinals."""
        hits = []
        for i, e in enumerate(bound):
            inner = e.child if isinstance(e, S.Alias) else e
            if isinstance(inner, S.Func) and inner.name.lower() in self._GENERATORS:
                hits.append((i, e, inner))
        if not hits:
            return None
        if len(hits) > 1:
            raise ResolutionError("only one generator (explode/posexplode) "
                                  "is allowed per SELECT list")
        i, e, fn = hits[0]
        lname = fn.name.lower()
        if lname == "stack":
            # stack(n, e1..ek) == inline(array(struct(row0...), ...)):
            # n rows of ceil(k/n) columns, missing cells NULL
            if not (fn.args and isinstance(fn.args[0], S.Literal)):
                raise ResolutionError("stack expects a literal row count")
            n_rows = int(fn.args[0].value)
            vals = fn.args[1:]
            width = -(-len(vals) // max(n_rows, 1))
            fields = tuple(T.StructField(f"col{c}", vals[c].dtype)
                           for c in range(width))
            st_t = T.StructType(fields)
            rows = []
            for r in range(n_rows):
                kv = []
                for c in range(width):
                    kv.append(S.Literal(f"col{c}", T.STRING))
                    idx = r * width + c
                    kv.append(vals[idx] if idx < len(vals)
                              else S.Cast(S.Literal(None, T.NULL),
                                          vals[c].dtype,
                                          dtype=vals[c].dtype))
                rows.append(S.Func("named_struct", kv, st_t))
            arr = S.Func("array", rows, T.ArrayType(st_t))
            fn = S.Func("inline", [arr], None)
            lname = "inline"
        arg_t = fn.args[0].dtype
        gen_cols: List[Tuple[str, T.DataType]] = []
        mode = ""
        if lname.startswith("inline"):
            if not (isinstance(arg_t, T.ArrayType)
                    and isinstance(arg_t.element, T.StructType)):
                raise ResolutionError("inline expects array<struct>")
            mode = "inline"
            gen_cols = [(f.name, f.dtype) for f in arg_t.element.fields]
        elif isinstance(arg_t, T.MapType):
            gen_cols = [("key", arg_t.key), ("value", arg_t.value)]
        elif isinstance(arg_t, T.ArrayType):
            col_name = e.name if isinstance(e, S.Alias) else "col"
            gen_cols = [(col_name, arg_t.element)]
        else:
            raise ResolutionError(f"{fn.name} expects an array or map "
                                  "argument")
        position = lname.startswith("posexplode")
        outer_gen = lname.endswith("_outer")
        g = S.Generate(input=child, gen=fn.args[0], outer=outer_gen,
                       position=position, mode=mode)
        nin = len(child.schema)
        g.schema = list(child.schema) \
            + ([("pos", T.I32)] if position else []) + gen_cols
        new_exprs: List[S.Expr] = []
        for j, b in enumerate(bound):
            if j != i:
                new_exprs.append(b)
            else:
                at = nin
                if position:
                    new_exprs.append(S.BoundRef(at, "pos", T.I32))
                    at += 1
                for nm, t in gen_cols:
                    new_exprs.append(S.BoundRef(at, nm, t))
                    at += 1
        out = S.Project(input=g, exprs=new_exprs)
        out.schema = [(_expr_name(x, k), x.dtype) for k, x in enumerate(new_exprs)]
        return out

    def _p_Distinct(self, p: S.Distinct, outer):
        child = self._plan(p.input, outer)
        out = S.Distinct(input=child)
        out.schema = child.schema
        return out

    def _p_Sample(self, p: S.Sample, outer):
        child = self._plan(p.input, outer)
        out = S.Sample(input=child, fraction=p.fraction, rows=p.rows, seed=p.seed)
        out.schema = child.schema
        return out

    def _p_Limit(self, p: S.Limit, outer):
        child = self._plan(p.input, outer)
        out = S.Limit(input=child, n=p.n, offset=p.offset)
        out.schema = child.schema
        return out

    def _p_Sort(self, p: S.Sort, outer):
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)
        if len(p.keys) == 1 and isinstance(p.keys[0].child, S.Col) \
                and p.keys[0].child.name == "__all__":
            # ORDER BY ALL: every output column, left to right
            asc = p.keys[0].ascending
            keys = [S.SortKey(S.BoundRef(i, n, t), asc, None)
                    for i, (n, t) in enumerate(child.schema)]
            out = S.Sort(input=child, keys=[self._expr(k, scope) if False else k
                                            for k in keys])
            for k in out.keys:
                k.dtype = k.child.dtype
            out.schema = child.schema
            return out
        keys: List[S.SortKey] = []
        extra_exprs: List[S.Expr] = []
        for k in p.keys:
            ke = self._resolve_sort_expr(k.child, child, scope, extra_exprs)
            keys.append(S.SortKey(ke, k.ascending, k.nulls_first))
        if extra_exprs and isinstance(child, S.Project):
            # widen the projection with hidden sort columns; Sort then projects back
            inner = child
            new_exprs = list(inner.exprs) + extra_exprs
            wide = S.Project(input=inner.input, exprs=new_exprs)
            wide.schema = [(_expr_name(e, i), e.dtype) for i, e in enumerate(new_exprs)]
            srt = S.Sort(input=wide, keys=keys)
            srt.schema = wide.schema
            trim = S.Project(input=srt, exprs=[
                S.BoundRef(i, n, t) for i, (n, t) in enumerate(inner.schema)])
            trim.schema = inner.schema
            return trim
        out = S.Sort(input=child, keys=keys)
        out.schema = child.schema
        return out

    def _resolve_sort_expr(self, e: S.Expr, child: S.Plan, scope: Scope, extra: List[S.Expr]) -> S.Expr:
        # ordinal sort key: ORDER BY 1
        if isinstance(e, S.Literal) and isinstance(e.value, int) and not isinstance(e.value, bool):
            idx = e.value - 1
            if 0 <= idx < len(child.schema):
                n, t = child.schema[idx]
                return S.BoundRef(idx, n, t)
            raise ResolutionError(
                f"ORDER BY position {e.value} is not in the select list "
                f"(1..{len(child.schema)})")
        # ORDER BY COUNT(*) / aggregate expressions over a grouped query
        if any(isinstance(x, S.AggFunc) for x in e.walk()) and isinstance(child, S.Project):
            agg = child.input
            if isinstance(agg, S.Filter):
                agg = agg.input
            if isinstance(agg, S.Aggregate):
                bound = self._bind_sort_agg(e, agg)
                idx = len(child.schema) + len(extra)
                extra.append(bound)
                return S.BoundRef(idx, f"__sort{idx}", bound.dtype)
        try:
            return self._expr(e, scope)
        except ResolutionError:
            if isinstance(e, S.Col) and e.qualifier:
                # ORDER BY t.col where the aggregate output column `col`
                # came from that key: qualifiers don't survive grouping,
                # so a uniquely-named output column matches (Spark allows
                # ORDER BY with the grouped key's qualified form)
                hits = [i for i, (n, _t) in enumerate(child.schema)
                        if n.lower() == e.name.lower()]
                if len(hits) == 1:
                    n, t = child.schema[hits[0]]
                    return S.BoundRef(hits[0], n, t)
            if isinstance(child, S.Project):
                inner_scope = self._child_scope(child.input, scope.outer)
                bound = self._expr(e, inner_scope)
                idx = len(child.schema) + len(extra)
                extra.append(bound)
                return S.BoundRef(idx, f"__sort{idx}", bound.dtype)
            raise

    def _bind_sort_agg(self, e: S.Expr, agg: S.Aggregate) -> S.Expr:
        """Bind a sort expression containing aggregates against an Aggregate
        node, appending missing aggregates to it."""
        inner_scope = self._child_scope(agg.input, None)
        ng = len(agg.group_by)

        def bind(x: S.Expr) -> S.Expr:
            if isinstance(x, S.AggFunc):
                bargs = [self._expr(a, inner_scope) for a in x.args
                         if not isinstance(a, S.Star)]
                name = _normalize_agg_name(x.name)
                rtype = agg_return_type(name, [a.dtype for a in bargs], x.distinct)
                bound = S.AggFunc(name, bargs, x.distinct, rtype, None)
                for ai, existing in enumerate(agg.aggs):
                    if repr(existing) == repr(bound):
                        return S.BoundRef(ng + ai, f"__agg{ai}", existing.dtype)
                agg.aggs.append(bound)
                agg.schema = list(agg.schema) + [(f"__agg{len(agg.aggs)-1}", rtype)]
                return S.BoundRef(ng + len(agg.aggs) - 1, f"__agg{len(agg.aggs)-1}", rtype)
            if isinstance(x, S.Col):
                # group key by name
                for gi in range(ng):
                    if agg.schema[gi][0].lower() == x.name.lower():
                        return S.BoundRef(gi, agg.schema[gi][0], agg.schema[gi][1])
                raise ResolutionError(f"cannot resolve {x.name} in sort over aggregate")
            ch = x.children()
            if not ch:
                return x
            return self._type_expr(x.with_children([bind(c) for c in ch]))

        return self._type_expr(bind(e))

    def _p_Aggregate(self, p: S.Aggregate, outer):
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)

        # expand * in projections (rare in aggregates: count(*) handled below)
        projections: List[S.Expr] = []
        for e in p.aggs:
            projections.extend(self._expand_star(e, scope) if isinstance(e, S.Star) else [e])

        # group-by keys: ordinals and alias references resolve against projections
        group_exprs: List[S.Expr] = []
        for g in p.group_by:
            if isinstance(g, S.Literal) and isinstance(g.value, int) and not isinstance(g.value, bool):
                idx = g.value - 1
                if not (0 <= idx < len(projections)):
                    raise ResolutionError(f"GROUP BY ordinal {g.value} out of range")
                tgt = projections[idx]
                group_exprs.append(tgt.child if isinstance(tgt, S.Alias) else tgt)
            elif isinstance(g, S.Col) and g.qualifier is None and not scope.find(g.name, None):
                # alias reference to a projection
                matched = None
                for pr in projections:
                    if isinstance(pr, S.Alias) and pr.name.lower() == g.name.lower():
                        matched = pr.child
                        break
                if matched is None:
                    raise ResolutionError(f"cannot resolve group key {g.name}")
                group_exprs.append(matched)
            else:
                group_exprs.append(g)
        bound_groups = [self._expr(g, scope) for g in group_exprs]

        # collect aggregate functions from projections
        agg_funcs: List[S.AggFunc] = []

        def bind_agg(e: S.Expr) -> S.Expr:
            """Bind a projection expr: AggFuncs -> refs into agg output;
            group exprs -> refs to group keys."""
            if isinstance(e, S.Alias):
                return S.Alias(bind_agg(e.child), e.name, None)
            if isinstance(e, (S.ScalarSubquery, S.Exists)):
                return self._expr(e, scope)
            if isinstance(e, S.InSubquery):
                sub = self._plan(e.plan, scope)
                return S.InSubquery(bind_agg(e.child), sub, e.negated, T.BOOL)
            # whole-expression matches a group key?
            for gi, (ge, be) in enumerate(zip(group_exprs, bound_groups)):
                if _expr_equal_unbound(e, ge):
                    return S.BoundRef(gi, _expr_name(e, gi), be.dtype)
            if isinstance(e, S.AggFunc):
                if e.name.lower() in ("grouping", "grouping_id"):
                    # positions of the referenced group keys (executor fills
                    # per-grouping-set values)
                    pos = []
                    for a in e.args:
                        for gi, ge in enumerate(group_exprs):
                            if _expr_equal_unbound(a, ge):
                                pos.append(gi)
                                break
                        else:
                            raise ResolutionError("grouping() argument must be a group key")
                    rt = T.I32 if e.name.lower() == "grouping" else T.I64
                    bound = S.AggFunc(e.name.lower(),
                                      [S.Literal(p_, T.I32) for p_ in pos], False, rt, None)
                    agg_funcs.append(bound)
                    return S.BoundRef(len(bound_groups) + len(agg_funcs) - 1,
                                      e.name.lower(), rt)
                if len(e.args) == 1 and isinstance(e.args[0], S.Star):
                    bargs = []
                else:
                    bargs = [self._expr(a, scope) for a in e.args]
                bfilter = _coerce_to_bool(self._expr(e.filter, scope)) if e.filter is not None else None
                name = _normalize_agg_name(e.name)
                rtype = agg_return_type(name, [a.dtype for a in bargs], e.distinct)
                bound = S.AggFunc(name, bargs, e.distinct, rtype, bfilter)
                pretty = _expr_name(e, 0)
                # dedup identical aggregates
                for ai, existing in enumerate(agg_funcs):
                    if repr(existing) == repr(bound):
                        return S.BoundRef(len(bound_groups) + ai, pretty, existing.dtype)
                agg_funcs.append(bound)
                ai = len(agg_funcs) - 1
                return S.BoundRef(len(bound_groups) + ai, pretty, rtype)
            ch = e.children()
            if not ch:
                if isinstance(e, S.Col):
                    # alias reference to a projection (HAVING n > 1)
                    if e.qualifier is None:
                        for pr in projections:
                            if isinstance(pr, S.Alias) and pr.name.lower() == e.name.lower() \
                                    and pr.child is not e:
                                return bind_agg(pr.child)
                    # column not in group keys — Spark errors; match that
                    hits = scope.find(e.name, e.qualifier)
                    if hits:
                        raise ResolutionError(
                            f"column {e.name} must appear in GROUP BY or inside an aggregate")
                    raise ResolutionError(f"cannot resolve column {e.name}")
                return e
            return self._type_expr(e.with_children([bind_agg(c) for c in ch]))

        final_exprs = [self._type_expr(bind_agg(e)) for e in projections]
        having_bound = None
        if p.having is not None:
            having_bound = _coerce_to_bool(self._type_expr(bind_agg(p.having)))

        agg = S.Aggregate(input=child, group_by=bound_groups, aggs=agg_funcs,
                          grouping_sets=p.grouping_sets)
        agg.schema = ([(_expr_name(g, i), g.dtype) for i, g in enumerate(bound_groups)]
                      + [(f"__agg{i}", a.dtype) for i, a in enumerate(agg_funcs)])
        top: S.Plan = agg
        if having_bound is not None:
            filt = S.Filter(input=agg, condition=having_bound)
            filt.schema = agg.schema
            top = filt
        proj = S.Project(input=top, exprs=final_exprs)
        proj.schema = [(_expr_name(e, i), e.dtype) for i, e in enumerate(final_exprs)]
        return proj

    def _p_Join(self, p: S.Join, outer):
        left = self._plan(p.left, outer)
        if p.right.__dict__.get("_lateral"):
            lat = self._lateral_project(p, left, outer)
            if lat is not None:
                return lat
        right = self._plan(p.right, outer)
        lfields = _scope_fields(left)
        rfields = _scope_fields(right)
        scope = Scope(lfields + rfields, outer)

        using = p.using
        if using and using == ["__natural__"]:
            lnames = {n.lower() for n, _ in left.schema}
            using = [n for n, _ in right.schema if n.lower() in lnames]
        on = None
        if using:
            conds = []
            for c in using:
                li = Scope(lfields).find(c, None)
                ri = Scope(rfields).find(c, None)
                if not li or not ri:
                    raise ResolutionError(f"USING column {c} not found on both sides")
                lref = S.BoundRef(li[0], c, lfields[li[0]].dtype)
                rref = S.BoundRef(len(lfields) + ri[0], c, rfields[ri[0]].dtype)
                cond = S.BinaryOp("=", lref, rref, T.BOOL)
                conds.append(cond)
            on = conds[0]
            for c in conds[1:]:
                on = S.BinaryOp("and", on, c, T.BOOL)
        elif p.on is not None:
            on = _coerce_to_bool(self._expr(p.on, scope))

        out = S.Join(left=left, right=right, how=p.how, on=on, using=using)
        if p.how in ("semi", "anti"):
            out.schema = list(left.schema)
            return out
        if p.how in ("rightsemi", "rightanti"):
            out.schema = list(right.schema)
            return out
        out.schema = list(left.schema) + list(right.schema)
        if using:
            # USING join: shared columns appear once. Normalized here to a
            # Project over the full-concat join so every later phase (prune,
            # reorder, executor) sees ONE coordinate system; key column comes
            # from the non-null side (right for RIGHT joins, coalesce for
            # FULL, left otherwise — Spark semantics).
            used = {c.lower() for c in using}
            nl = len(left.schema)
            rpos = {n.lower(): j for j, (n, t) in enumerate(right.schema)}
            exprs: List[S.Expr] = []
            schema = []
            for i, (n, t) in enumerate(left.schema):
                if n.lower() in used:
                    lref = S.BoundRef(i, n, t)
                    rref = S.BoundRef(nl + rpos[n.lower()], n,
                                      right.schema[rpos[n.lower()]][1])
                    if p.how == "right":
                        e = rref
                    elif p.how == "full":
                        e = S.Func("coalesce", [lref, rref], dtype=t)
                    else:
                        e = lref
                    e = S.Alias(e, n, t) if not isinstance(e, S.BoundRef) else e
                    exprs.append(e)
                else:
                    exprs.append(S.BoundRef(i, n, t))
                schema.append((n, t))
            for j, (n, t) in enumerate(right.schema):
                if n.lower() not in used:
                    exprs.append(S.BoundRef(nl + j, n, t))
                    schema.append((n, t))
            proj = S.Project(input=out, exprs=exprs)
            proj.schema = schema
            # preserve per-column qualifiers for parent join scopes
            concat = lfields + rfields
            ovr = []
            for e, (n, t) in zip(exprs, schema):
                q = concat[e.index].qualifier if isinstance(e, S.BoundRef) else None
                ovr.append(Field(n, t, q))
            proj.__dict__["_scope_fields_override"] = ovr
            return proj
        return out

    def _p_SetOp(self, p: S.SetOp, outer):
        left = self._plan(p.left, outer)
        right = self._plan(p.right, outer)
        if len(left.schema) != len(right.schema):
            raise ResolutionError("set operation inputs have different column counts")
        schema = []
        for (ln, lt), (rn, rt) in zip(left.schema, right.schema):
            schema.append((ln, T.common_type(lt, rt)))
        out = S.SetOp(op=p.op, left=left, right=right, is_all=p.is_all, by_name=p.by_name)
        out.schema = schema
        return out

    def _p_WindowPlan(self, p: S.WindowPlan, outer):
        child = self._plan(p.input, outer)
        scope = self._child_scope(child, outer)
        bound = [self._expr(e, scope) for e in p.window_exprs]
        out = S.WindowPlan(input=child, window_exprs=bound)
        out.schema = list(child.schema) + [(_expr_name(e, len(child.schema) + i), e.dtype)
                                           for i, e in enumerate(bound)]
        return out

    # commands ------------------------------------------------------------
    def _p_CreateView(self, p: S.CreateView, outer):
        body = self._plan(p.input, outer)
        out = S.CreateView(name=p.name, input=body, replace=p.replace, temporary=p.temporary)
        out.schema = []
        return out

    def _p_CreateTable(self, p: S.CreateTable, outer):
        inp = self._plan(p.input, outer) if p.input is not None else None
        out = S.CreateTable(name=p.name, columns=p.columns, input=inp, format=p.format,
                            location=p.location, replace=p.replace,
                            if_not_exists=p.if_not_exists, options=p.options)
        out.schema = []
        return out

    def _p_DropTable(self, p: S.DropTable, outer):
        p.schema = []
        return p

    def _lateral_project(self, p: S.Join, left, outer):
        """LATERAL (SELECT exprs) x — the projection case: inline the
        lateral's expressions as extra columns computed over the left side
        (the reference's DecorrelateLateralProjection,
        sail-logical-optimizer). Returns None for shapes that need a real
        correlated join (lateral body with a FROM)."""
        node = p.right
        alias, col_aliases = None, None
        if isinstance(node, S.SubqueryAlias):
            alias, col_aliases = node.alias, node.column_aliases
            node = node.input
        if not (isinstance(node, S.Project)
                and isinstance(node.input, S.LocalRelation)
                and p.how in ("inner", "cross")):
            return None
        lscope = self._child_scope(left, outer)
        exprs = []
        for i, (n, _t) in enumerate(left.schema):
            exprs.append(S.BoundRef(i, n, _t))
        lat_exprs = []
        for i, e in enumerate(node.exprs):
            b = self._expr(e, lscope)
            nm = (col_aliases[i] if col_aliases and i < len(col_aliases)
                  else _expr_name(e, i))
            lat_exprs.append(S.Alias(b, nm, b.dtype))
        pr = S.Project(input=left, exprs=exprs + lat_exprs)
        pr.schema = list(left.schema) + [(a.name, a.dtype)
                                         for a in lat_exprs]
        # left columns keep their own qualifiers; lateral columns get the
        # subquery alias (x.col addressing)
        pr.__dict__["_scope_fields_override"] = _scope_fields(left) + [
            Field(a.name, a.dtype, alias) for a in lat_exprs]
        return pr

    def _insert_column_align(self, inp, cols, tgt):
        """INSERT INTO t (a, c): align the input's columns to the target
        schema — named columns by position in the list, the rest NULL."""
        names = [n.lower() for n, _t in tgt]
        pos = {}
        for i, c in enumerate(cols):
            if c.lower() not in names:
                raise ResolutionError(
                    f"INSERT column {c} not in target {names}")
            pos[c.lower()] = i
        if len(inp.schema) != len(cols):
            raise ResolutionError(
                f"INSERT specifies {len(cols)} columns but query produces "
                f"{len(inp.schema)}")
        exprs = []
        for tn, tt in tgt:
            i = pos.get(tn.lower())
            if i is None:
                exprs.append(S.Alias(
                    S.Cast(S.Literal(None, T.NULL), tt, dtype=tt), tn, tt))
            else:
                sn, st = inp.schema[i]
                exprs.append(S.Alias(S.BoundRef(i, sn, st), tn, st))
        pr = S.Project(input=inp, exprs=exprs)
        pr.schema = [(e.name, e.dtype) for e in exprs]
        return pr

    def _p_InsertInto(self, p: S.InsertInto, outer):
        inp = self._plan(p.input, outer)
        if p.columns:
            tgt = None
            head0, _, rest0 = p.table.partition(".")
            if head0.lower() in ("parquet", "csv", "json", "delta",
                                 "iceberg") and rest0:
                from ..datasource.registry import infer_source_schema

                tgt = infer_source_schema(head0.lower(), [rest0], {})
            else:
                tgt = self.catalog.table_schema(p.table)
            if tgt is None:
                raise ResolutionError(f"table not found: {p.table}")
            inp = self._insert_column_align(inp, p.columns, list(tgt))
        head, _, rest = p.table.partition(".")
        if head.lower() in ("parquet", "csv", "json", "delta", "iceberg") and rest:
            # INSERT INTO delta.`/path` -> datasource append/overwrite;
            # positional semantics: rename/cast the input to the target
            # schema (VALUES rows arrive as col1/col2/...)
            from ..datasource.registry import infer_source_schema

            tgt = infer_source_schema(head.lower(), [rest], {})
            if len(inp.schema) != len(tgt):
                raise ResolutionError(
                    f"INSERT INTO {p.table}: {len(inp.schema)} columns for "
                    f"{len(tgt)}-column target")
            exprs = []
            for i, ((sn, st), (tn, tt)) in enumerate(zip(inp.schema, tgt)):
                ref = S.BoundRef(i, sn, st)
                e = S.Cast(ref, tt, dtype=tt) if st != tt else ref
                exprs.append(S.Alias(e, tn, tt))
            pr = S.Project(input=inp, exprs=exprs)
            pr.schema = list(tgt)
            out = S.Write(input=pr, format=head.lower(), path=rest,
                          mode="overwrite" if p.overwrite else "append")
            out.schema = []
            return out
        if self.catalog.table_schema(p.table) is None:
            raise ResolutionError(f"table not found: {p.table}")
        out = S.InsertInto(table=p.table, input=inp, overwrite=p.overwrite)
        out.schema = []
        return out

    def _p_Write(self, p: S.Write, outer):
        inp = self._plan(p.input, outer)
        out = S.Write(input=inp, format=p.format, path=p.path, table=p.table,
                      mode=p.mode, partition_by=p.partition_by, options=p.options)
        out.schema = []
        return out

    def _target_schema(self, name: str):
        head, _, rest = name.partition(".")
        if head.lower() == "delta" and rest:
            from ..datasource.delta import infer_schema

            return infer_schema([rest]), ("delta", rest)
        if head.lower() == "iceberg" and rest:
            from ..datasource.iceberg import infer_schema

            return infer_schema([rest]), ("iceberg", rest)
        schema = self.catalog.table_schema(name)
        if schema is None:
            raise ResolutionError(f"table not found: {name}")
        return schema, ("catalog", name)

    def _p_MergeInto(self, p: S.MergeInto, outer):
        tschema, tref = self._target_schema(p.target)
        source = self._plan(p.source, outer)
        tq = p.target_alias or p.target.split(".")[-1]
        sq = p.source_alias or _plan_qualifier(source)
        fields = [Field(n, t, tq) for n, t in tschema] +                  [Field(n, t, sq) for n, t in source.schema]
        scope = Scope(fields, outer)
        on = _coerce_to_bool(self._expr(p.on, scope))

        def bind_action(a: S.MergeAction, star_schema=None) -> S.MergeAction:
            cond = _coerce_to_bool(self._expr(a.condition, scope)) if a.condition is not None else None
            if a.kind == "update_star":
                assigns = []
                snames = {n.lower(): i for i, (n, _) in enumerate(source.schema)}
                for ti, (n, t) in enumerate(tschema):
                    if n.lower() in snames:
                        si = snames[n.lower()]
                        assigns.append((n, S.BoundRef(len(tschema) + si, n, source.schema[si][1])))
                return S.MergeAction("update", cond, assigns)
            if a.kind == "insert_star":
                cols, vals = [], []
                snames = {n.lower(): i for i, (n, _) in enumerate(source.schema)}
                for n, t in tschema:
                    if n.lower() in snames:
                        si = snames[n.lower()]
                        cols.append(n)
                        vals.append(S.BoundRef(len(tschema) + si, n, source.schema[si][1]))
                return S.MergeAction("insert", cond, insert_columns=cols, insert_values=vals)
            if a.kind == "update":
                assigns = []
                for name, e in a.assignments:
                    col = name.split(".")[-1]
                    if not any(col.lower() == n.lower() for n, _ in tschema):
                        raise ResolutionError(f"MERGE SET column {col} not in target")
                    assigns.append((col, self._expr(e, scope)))
                return S.MergeAction("update", cond, assigns)
            if a.kind == "insert":
                vals = [self._expr(e, scope) for e in a.insert_values]
                return S.MergeAction("insert", cond, insert_columns=a.insert_columns,
                                     insert_values=vals)
            return S.MergeAction(a.kind, cond)

        out = S.MergeInto(target=p.target, target_alias=p.target_alias, source=source,
                          source_alias=p.source_alias, on=on,
                          matched=[bind_action(a) for a in p.matched],
                          not_matched=[bind_action(a) for a in p.not_matched],
                          not_matched_by_source=[bind_action(a) for a in p.not_matched_by_source])
        out.schema = []
        out.__dict__["_target_schema"] = tschema
        out.__dict__["_target_ref"] = tref
        return out

    def _p_UpdateTable(self, p: S.UpdateTable, outer):
        tschema, tref = self._target_schema(p.table)
        scope = Scope([Field(n, t, p.table.split(".")[-1]) for n, t in tschema], outer)
        assigns = []
        for name, e in p.assignments:
            col = name.split(".")[-1]
            assigns.append((col, self._expr(e, scope)))
        cond = _coerce_to_bool(self._expr(p.condition, scope)) if p.condition is not None else None
        out = S.UpdateTable(table=p.table, assignments=assigns, condition=cond)
        out.schema = []
        out.__dict__["_target_schema"] = tschema
        out.__dict__["_target_ref"] = tref
        return out

    def _p_DeleteFrom(self, p: S.DeleteFrom, outer):
        tschema, tref = self._target_schema(p.table)
        scope = Scope([Field(n, t, p.table.split(".")[-1]) for n, t in tschema], outer)
        cond = _coerce_to_bool(self._expr(p.condition, scope)) if p.condition is not None else None
        out = S.DeleteFrom(table=p.table, condition=cond)
        out.schema = []
        out.__dict__["_target_schema"] = tschema
        out.__dict__["_target_ref"] = tref
        return out

    def _p_Explain(self, p: S.Explain, outer):
        inp = self._plan(p.input, outer)
        out = S.Explain(input=inp, mode=p.mode)
        out.schema = [("plan", T.STRING)]
        return out

    def _p_SetConfig(self, p: S.SetConfig, outer):
        p.schema = [("key", T.STRING), ("value", T.STRING)]
        return p

    def _p_ShowTables(self, p: S.ShowTables, outer):
        p.schema = [("namespace", T.STRING), ("tableName", T.STRING), ("isTemporary", T.BOOL)]
        return p

    def _p_VacuumTable(self, p: S.VacuumTable, outer):
        p.schema = [("removed_file", T.STRING)]
        return p

    def _p_DescribeHistory(self, p: S.DescribeHistory, outer):
        p.schema = [("version", T.I64), ("timestamp_ms", T.I64),
                    ("operation", T.STRING), ("num_added_files", T.I64),
                    ("num_removed_files", T.I64)]
        return p

    def _p_AlterTable(self, p: S.AlterTable, outer):
        if self.catalog.table_schema(p.name) is None:
            raise ResolutionError(f"table not found: {p.name}")
        p.schema = [("result", T.STRING)]
        return p

    def _p_ShowFunctions(self, p: S.ShowFunctions, outer):
        p.schema = [("function", T.STRING)]
        return p

    def _p_ShowDatabases(self, p: S.ShowDatabases, outer):
        p.schema = [("namespace", T.STRING)]
        return p

    def _p_CacheTable(self, p: S.CacheTable, outer):
        inp = self._plan(p.input, outer) if p.input is not None else None
        if inp is None:
            # cache an existing view/table by name
            inp = self._plan(S.Read(table=p.name), outer)
        out = S.CacheTable(name=p.name, input=inp)
        out.schema = [("result", T.STRING)]
        return out

    def _p_UncacheTable(self, p: S.UncacheTable, outer):
        p.schema = [("result", T.STRING)]
        return p

    def _p_AnalyzeTable(self, p: S.AnalyzeTable, outer):
        if self.catalog.table_schema(p.name) is None \
                and getattr(self.catalog, "view_plan", lambda n: None)(p.name) is None:
            raise ResolutionError(f"table not found: {p.name}")
        p.schema = [("column", T.STRING), ("rows", T.I64), ("ndv", T.I64)]
        return p

    def _p_DescribeTable(self, p: S.DescribeTable, outer):
        p.schema = [("col_name", T.STRING), ("data_type", T.STRING), ("comment", T.STRING)]
        return p

    def _p_DescribeQuery(self, p: S.DescribeQuery, outer):
        inp = self._plan(p.input, outer)
        out = S.DescribeQuery(input=inp)
        out.schema = [("col_name", T.STRING), ("data_type", T.STRING),
                      ("comment", T.STRING)]
        return out

    def _p_ShowColumns(self, p: S.ShowColumns, outer):
        p.schema = [("col_name", T.STRING)]
        return p

    def _p_ShowCreateTable(self, p: S.ShowCreateTable, outer):
        p.schema = [("createtab_stmt", T.STRING)]
        return p

    def _p_ShowViews(self, p: S.ShowViews, outer):
        p.schema = [("namespace", T.STRING), ("viewName", T.STRING),
                    ("isTemporary", T.BOOL)]
        return p

    def _p_ShowPartitions(self, p: S.ShowPartitions, outer):
        p.schema = [("partition", T.STRING)]
        return p

    def _p_ShowTblProperties(self, p: S.ShowTblProperties, outer):
        p.schema = [("key", T.STRING), ("value", T.STRING)]
        return p

    def _p_ShowCatalogs(self, p: S.ShowCatalogs, outer):
        p.schema = [("catalog", T.STRING)]
        return p

    def _p_UseDatabase(self, p: S.UseDatabase, outer):
        p.schema = []
        return p

    def _p_CreateDatabase(self, p: S.CreateDatabase, outer):
        p.schema = []
        return p

    def _p_DropDatabase(self, p: S.DropDatabase, outer):
        p.schema = []
        return p

    def _p_RefreshTable(self, p: S.RefreshTable, outer):
        p.schema = [("result", T.STRING)]
        return p

    def _p_TruncateTable(self, p: S.TruncateTable, outer):
        if self.catalog.table_schema(p.name) is None:
            raise ResolutionError(f"table not found: {p.name}")
        p.schema = []
        return p

    def _p_CommentOn(self, p: S.CommentOn, outer):
        p.schema = []
        return p

    # =====================================================================
    # expressions
    # =====================================================================
    def _child_scope(self, child: S.Plan, outer) -> Scope:
        return Scope(_scope_fields(child), outer)

    def _expand_star(self, e: S.Expr, scope: Scope) -> List[S.Expr]:
        if isinstance(e, S.Star):
            out = []
            for i, f in enumerate(scope.fields):
                if e.qualifier and (f.qualifier or "").lower() != e.qualifier.lower():
                    continue
                out.append(S.BoundRef(i, f.name, f.dtype))
            if not out:
                raise ResolutionError(f"star expansion found no columns for {e.qualifier}")
            return out
        return [e]

    def _expr(self, e: S.Expr, scope: Scope) -> S.Expr:
        if e is None:
            return None
        if isinstance(e, S.Literal):
            out = S.Literal(e.value, e.dtype or _infer_literal_type(e.value))
            return out
        if isinstance(e, S.BoundRef):
            return e
        if isinstance(e, S.Col):
            hits = scope.find(e.name, e.qualifier)
            if len(hits) == 1:
                f = scope.fields[hits[0]]
                return S.BoundRef(hits[0], f.name, f.dtype)
            if len(hits) > 1:
                # same qualifier on every hit = genuine duplicate column set
                # (e.g. SELECT * over a self-join projection): take first;
                # different qualifiers = ambiguous (Spark AMBIGUOUS_REFERENCE)
                f0 = scope.fields[hits[0]]
                quals = {(scope.fields[h].qualifier or "") for h in hits}
                if len(quals) == 1 and all(scope.fields[h].dtype == f0.dtype for h in hits):
                    return S.BoundRef(hits[0], f0.name, f0.dtype)
                raise ResolutionError(
                    f"ambiguous column {e.name} (candidates: "
                    + ", ".join(sorted(f"{scope.fields[h].qualifier or '?'}.{e.name}"
                                       for h in hits)) + ")")
            # outer scope (correlated subquery)
            s = scope.outer
            depth = 0
            while s is not None:
                hits = s.find(e.name, e.qualifier)
                if hits:
                    if depth > 0:
                        raise ResolutionError("correlation beyond one level not supported")
                    f = s.fields[hits[0]]
                    return S.OuterRef(hits[0], f.name, f.dtype)
                s = s.outer
                depth += 1
            raise ResolutionError(f"cannot resolve column "
                                  f"{(e.qualifier + '.') if e.qualifier else ''}{e.name}")
        if isinstance(e, S.Alias):
            c = self._expr(e.child, scope)
            return S.Alias(c, e.name, c.dtype)
        if isinstance(e, S.ScalarSubquery):
            sub = self._plan(e.plan, scope)
            if len(sub.schema) != 1:
                raise ResolutionError("scalar subquery must return one column")
            return S.ScalarSubquery(plan=sub, dtype=sub.schema[0][1])
        if isinstance(e, S.InSubquery):
            c = self._expr(e.child, scope)
            sub = self._plan(e.plan, scope)
            return S.InSubquery(c, sub, e.negated, T.BOOL)
        if isinstance(e, S.Exists):
            sub = self._plan(e.plan, scope)
            return S.Exists(plan=sub, negated=e.negated, dtype=T.BOOL)
        if isinstance(e, S.AggFunc):
            bargs = [self._expr(a, scope) for a in e.args if not isinstance(a, S.Star)]
            name = _normalize_agg_name(e.name)
            rtype = agg_return_type(name, [a.dtype for a in bargs], e.distinct)
            bfilter = _coerce_to_bool(self._expr(e.filter, scope)) if e.filter is not None else None
            return S.AggFunc(name, bargs, e.distinct, rtype, bfilter)
        if isinstance(e, S.WindowExpr):
            f = self._expr_window_func(e.func, scope)
            part = [self._expr(x, scope) for x in e.partition_by]
            order = [S.SortKey(self._expr(k.child, scope), k.ascending, k.nulls_first)
                     for k in e.order_by]
            return S.WindowExpr(func=f, partition_by=part, order_by=order,
                                frame=e.frame, dtype=f.dtype)
        if isinstance(e, S.Func) and e.name.lower() in ("if", "iff") \
                and len(e.args) == 3:
            # IF(c, a, b) == CASE WHEN c THEN a ELSE b END — route through
            # the CASE machinery (handles strings/structs/dicts correctly)
            return self._expr(S.CaseWhen(branches=[(e.args[0], e.args[1])],
                                         else_=e.args[2]), scope)
        if isinstance(e, S.Func) and e.name.lower() in (
                "make_dt_interval", "make_ym_interval", "make_interval",
                "try_make_interval"):
            # intervals are literal-folded: the engine stores day-time
            # intervals as ("__interval__", months, micros) literal tuples
            vals = []
            for a in e.args:
                lit = a
                if isinstance(lit, S.Cast):
                    lit = lit.child
                if isinstance(lit, S.UnaryOp) and lit.op == "neg" \
                        and isinstance(lit.child, S.Literal):
                    vals.append(-lit.child.value)
                elif isinstance(lit, S.Literal):
                    vals.append(lit.value)
                else:
                    raise ResolutionError(
                        f"{e.name} requires literal arguments")
            if e.name.lower() in ("make_interval", "try_make_interval"):
                # (years, months, weeks, days, hours, mins, secs)
                vals += [0] * (7 - len(vals))
                months = int(vals[0]) * 12 + int(vals[1])
                micros = ((int(vals[2]) * 7 + int(vals[3])) * 86_400_000_000
                          + int(vals[4]) * 3_600_000_000
                          + int(vals[5]) * 60_000_000
                          + int(round(float(vals[6]) * 1_000_000)))
                return S.Literal((_INTERVAL, months, micros), T.NULL)
            vals += [0] * (4 - len(vals))
            if e.name.lower() == "make_ym_interval":
                return S.Literal((_INTERVAL, int(vals[0]) * 12 + int(vals[1]),
                                  0), T.NULL)
            micros = (int(vals[0]) * 86_400_000_000
                      + int(vals[1]) * 3_600_000_000
                      + int(vals[2]) * 60_000_000
                      + int(round(float(vals[3]) * 1_000_000)))
            return S.Literal((_INTERVAL, 0, micros), T.NULL)
        if isinstance(e, S.Func) and any(isinstance(a, S.Lambda) for a in e.args):
            return self._resolve_hof(e, scope)
        # generic: resolve children then type
        ch = [self._expr(c, scope) for c in e.children()]
        out = e.with_children(ch) if ch else copy.copy(e)
        return self._type_expr(out)

    _HOF = {"transform", "filter", "exists", "forall", "array_filter",
            "zip_with", "aggregate", "reduce", "map_zip_with",
            "transform_keys", "transform_values", "map_filter",
            "array_sort"}

    def _resolve_hof(self, e: S.Func, scope: Scope) -> S.Expr:
        """Higher-order array functions with lambdas (ref: sail-plan
        resolver/expression/lambda.rs role). The lambda body is bound with
        params at BoundRef 0..k-1 and the enclosing row's columns at k+i —
        the evaluator builds a flattened-element chunk in that layout."""
        name = e.name.lower()
        if name not in self._HOF:
            raise ResolutionError(f"unsupported lambda function {e.name}")
        if name == "zip_with":
            return self._resolve_zip_with(e, scope)
        if name in ("aggregate", "reduce"):
            return self._resolve_reduce(e, scope)
        if name == "map_zip_with":
            m1 = self._expr(e.args[0], scope)
            m2 = self._expr(e.args[1], scope)
            if not (isinstance(m1.dtype, T.MapType)
                    and isinstance(m2.dtype, T.MapType)):
                raise ResolutionError("map_zip_with expects two maps")
            blam = self._bind_lambda(
                e.args[2], [m1.dtype.key, m1.dtype.value, m2.dtype.value],
                scope)
            return S.Func(name, [m1, m2, blam],
                          T.MapType(m1.dtype.key, blam.dtype))
        if name in ("transform_keys", "transform_values", "map_filter"):
            m = self._expr(e.args[0], scope)
            if not isinstance(m.dtype, T.MapType):
                raise ResolutionError(f"{name} expects a map argument")
            blam = self._bind_lambda(e.args[1], [m.dtype.key, m.dtype.value], scope)
            if name == "map_filter":
                blam = S.Lambda(blam.params, _coerce_to_bool(blam.body), T.BOOL)
                t = m.dtype
            elif name == "transform_keys":
                t = T.MapType(blam.dtype, m.dtype.value)
            else:
                t = T.MapType(m.dtype.key, blam.dtype)
            return S.Func(name, [m, blam], t)
        if name == "array_sort" and len(e.args) > 1:
            arr = self._expr(e.args[0], scope)
            if not isinstance(arr.dtype, T.ArrayType):
                raise ResolutionError("array_sort expects an array")
            et = arr.dtype.element
            blam = self._bind_lambda(e.args[1], [et, et], scope)
            return S.Func(name, [arr, blam], arr.dtype)
        arr = self._expr(e.args[0], scope)
        if not isinstance(arr.dtype, T.ArrayType):
            raise ResolutionError(f"{name} expects an array argument")
        lam = e.args[1]
        if not isinstance(lam, S.Lambda):
            raise ResolutionError(f"{name} expects a lambda")
        k = len(lam.params)
        if k > 2:
            raise ResolutionError("lambdas take at most (element, index)")
        ptypes = [arr.dtype.element] + ([T.I32] if k == 2 else [])
        lam_fields = [Field(lam.params[i], ptypes[i]) for i in range(k)]
        lscope = Scope(lam_fields + list(scope.fields), scope.outer)
        body = self._expr(lam.body, lscope)
        if name in ("filter", "exists", "forall", "array_filter"):
            body = _coerce_to_bool(body)
        blam = S.Lambda(lam.params, body, body.dtype)
        if name == "transform":
            t = T.ArrayType(body.dtype)
        elif name in ("filter", "array_filter"):
            t = arr.dtype
        else:
            t = T.BOOL
        return S.Func("filter" if name == "array_filter" else name,
                      [arr, blam], t)

    def _bind_lambda(self, lam, ptypes, scope):
        if not isinstance(lam, S.Lambda):
            raise ResolutionError("expected a lambda")
        if len(lam.params) != len(ptypes):
            raise ResolutionError(
                f"lambda expects {len(ptypes)} parameters, got {len(lam.params)}")
        fields = [Field(lam.params[i], ptypes[i]) for i in range(len(ptypes))]
        body = self._expr(lam.body, Scope(fields + list(scope.fields), scope.outer))
        return S.Lambda(lam.params, body, body.dtype)

    def _resolve_zip_with(self, e, scope):
        a = self._expr(e.args[0], scope)
        b = self._expr(e.args[1], scope)
        if not (isinstance(a.dtype, T.ArrayType) and isinstance(b.dtype, T.ArrayType)):
            raise ResolutionError("zip_with expects two arrays")
        blam = self._bind_lambda(e.args[2], [a.dtype.element, b.dtype.element], scope)
        return S.Func("zip_with", [a, b, blam], T.ArrayType(blam.dtype))

    def _resolve_reduce(self, e, scope):
        """aggregate(arr, init, (acc, x) -> merge [, acc -> finish])"""
        arr = self._expr(e.args[0], scope)
        if not isinstance(arr.dtype, T.ArrayType):
            raise ResolutionError("aggregate expects an array")
        init = self._expr(e.args[1], scope)
        acc_t = init.dtype
        merge = self._bind_lambda(e.args[2], [acc_t, arr.dtype.element], scope)
        if merge.dtype != acc_t:
            # accumulator type widened by the merge (e.g. int init, double
            # body): rebind once with the widened type
            acc_t = merge.dtype
            merge = self._bind_lambda(e.args[2], [acc_t, arr.dtype.element], scope)
        out_t = acc_t
        args = [arr, init, merge]
        if len(e.args) > 3:
            fin = self._bind_lambda(e.args[3], [acc_t], scope)
            out_t = fin.dtype
            args.append(fin)
        return S.Func("aggregate", args, out_t)

    def _expr_window_func(self, f: S.Expr, scope: Scope) -> S.Expr:
        if isinstance(f, S.AggFunc):
            return self._expr(f, scope)
        if isinstance(f, S.Func):
            args = [self._expr(a, scope) for a in f.args]
            name = f.name.lower()
            if name in ("row_number", "rank", "dense_rank", "ntile"):
                t = T.I32
            elif name in ("percent_rank", "cume_dist"):
                t = T.F64
            elif name in ("lag", "lead", "nth_value"):
                t = args[0].dtype if args else T.NULL
            else:
                t = scalar_return_type(name, [a.dtype for a in args]) or T.NULL
            return S.Func(name, args, t)
        raise ResolutionError(f"unsupported window function {f!r}")

    # -- typing rules ------------------------------------------------------
    def _type_expr(self, e: S.Expr) -> S.Expr:
        if e.dtype is not None and not isinstance(e, (S.BinaryOp, S.UnaryOp, S.Func, S.Cast,
                                                      S.CaseWhen, S.InList, S.Between, S.Like)):
            return e
        if isinstance(e, S.BinaryOp):
            return _type_binary(e)
        if isinstance(e, S.UnaryOp):
            if e.op == "not":
                e.child = _coerce_to_bool(e.child)
                e.dtype = T.BOOL
            elif e.op in ("isnull", "isnotnull"):
                e.dtype = T.BOOL
            elif e.op == "neg":
                e.dtype = e.child.dtype
            return e
        if isinstance(e, S.Cast):
            e.dtype = e.to
            return e
        if isinstance(e, S.CaseWhen):
            e.branches = [(_coerce_to_bool(c), v) for c, v in e.branches]
            ts = [v.dtype for _, v in e.branches] + ([e.else_.dtype] if e.else_ is not None else [])
            t = ts[0]
            for x in ts[1:]:
                t = T.common_type(t, x)
            e.dtype = t
            return e
        if isinstance(e, (S.InList, S.Between, S.Like, S.Exists, S.InSubquery)):
            e.dtype = T.BOOL
            return e
        if isinstance(e, S.SortKey):
            e.dtype = e.child.dtype
            return e
        if isinstance(e, S.Func):
            t = scalar_return_type(e.name, [a.dtype for a in e.args])
            if t is None:
                udf = getattr(self.catalog, "udf", None)
                info = udf(e.name) if udf else None
                if info is not None:
                    e.dtype = info[1]
                    e.__dict__["_is_udf"] = True
                    return e
                raise ResolutionError(f"unknown function {e.name}")
            # structural return types
            if e.name == "from_avro":
                # type = struct derived from the literal json schema arg
                if len(e.args) > 1 and isinstance(e.args[1], S.Literal):
                    import json as _json

                    from ..engine.functions_ext import _struct_type_from_avro

                    t = _struct_type_from_avro(_json.loads(e.args[1].value))
            if e.name == "from_protobuf":
                # type = struct from the literal (messageName, descFile)
                if len(e.args) > 2 and isinstance(e.args[1], S.Literal) \
                        and isinstance(e.args[2], S.Literal):
                    from ..engine.functions_ext import (_pb_class,
                                                        _pb_struct_type)

                    cls = _pb_class(str(e.args[1].value),
                                    str(e.args[2].value))
                    t = _pb_struct_type(cls.DESCRIPTOR)
            if e.name in ("variant_get", "try_variant_get") \
                    and len(e.args) > 2 and isinstance(e.args[2], S.Literal):
                try:
                    t = T.type_from_name(str(e.args[2].value))
                except ValueError:
                    pass
            if e.name == "coalesce" or e.name in ("nvl", "ifnull"):
                tt = e.args[0].dtype
                for a in e.args[1:]:
                    tt = T.common_type(tt, a.dtype)
                t = tt
            elif e.name == "struct":
                fields = []
                for i, a in enumerate(e.args):
                    nm = _expr_name(a, i)
                    fields.append(T.StructField(nm, a.dtype))
                t = T.StructType(tuple(fields))
            elif e.name == "named_struct":
                fields = []
                for i in range(0, len(e.args), 2):
                    k = e.args[i]
                    if not isinstance(k, S.Literal):
                        raise ResolutionError("named_struct expects literal field names")
                    fields.append(T.StructField(str(k.value), e.args[i + 1].dtype))
                t = T.StructType(tuple(fields))
            elif e.name == "arrays_zip":
                fields = []
                for i, a in enumerate(e.args):
                    if not isinstance(a.dtype, T.ArrayType):
                        raise ResolutionError("arrays_zip expects arrays")
                    nm = a.name if isinstance(a, (S.BoundRef, S.Alias)) else str(i)
                    fields.append(T.StructField(nm, a.dtype.element))
                t = T.ArrayType(T.StructType(tuple(fields)))
            elif e.name == "map_entries":
                mt = e.args[0].dtype
                if not isinstance(mt, T.MapType):
                    raise ResolutionError("map_entries expects a map")
                t = T.ArrayType(T.StructType((T.StructField("key", mt.key),
                                              T.StructField("value", mt.value))))
            elif e.name == "map_from_entries":
                at = e.args[0].dtype
                if not (isinstance(at, T.ArrayType)
                        and isinstance(at.element, T.StructType)
                        and len(at.element.fields) == 2):
                    raise ResolutionError(
                        "map_from_entries expects array<struct<k,v>>")
                t = T.MapType(at.element.fields[0].dtype,
                              at.element.fields[1].dtype)
            elif e.name in ("from_json", "from_csv", "from_xml"):
                if len(e.args) < 2 or not isinstance(e.args[1], S.Literal):
                    raise ResolutionError(f"{e.name} expects a literal schema string")
                from ..sql.parser import parse_ddl_schema

                fields = parse_ddl_schema(str(e.args[1].value))
                t = T.StructType(tuple(T.StructField(n, ft) for n, ft in fields))
            elif e.name == "get_field":
                st = e.args[0].dtype
                fname = e.args[1].value if isinstance(e.args[1], S.Literal) else None
                if isinstance(st, T.StructType) and fname is not None:
                    ft = None
                    for f in st.fields:
                        if f.name.lower() == str(fname).lower():
                            ft = f.dtype
                    if ft is None:
                        raise ResolutionError(f"no field {fname} in {st!r}")
                    t = ft
                else:
                    raise ResolutionError("get_field expects struct.fieldname")
            e.dtype = t
            return e
        if isinstance(e, S.Alias):
            e.dtype = e.child.dtype
            return e
        if e.dtype is None:
            raise ResolutionError(f"cannot type expression {e!r}")
        return e


# ---------------------------------------------------------------------------

_INTERVAL = "__interval__"


def _is_interval_lit(e: S.Expr) -> bool:
    return isinstance(e, S.Literal) and isinstance(e.value, tuple) and len(e.value) == 3 \
        and e.value[0] == _INTERVAL


def _type_binary(e: S.BinaryOp) -> S.Expr:
    lt, rt = e.left.dtype, e.right.dtype
    if e.op in ("and", "or"):
        e.left = _coerce_to_bool(e.left)
        e.right = _coerce_to_bool(e.right)
        e.dtype = T.BOOL
        return e
    # date/timestamp +- interval
    if e.op in ("+", "-") and (_is_interval_lit(e.right) or _is_interval_lit(e.left)):
        ivl = e.right if _is_interval_lit(e.right) else e.left
        other = e.left if ivl is e.right else e.right
        _, months, micros = ivl.value
        sign = -1 if e.op == "-" else 1
        if isinstance(other.dtype, T.DateType):
            out = other
            if months:
                out = S.Func("add_months", [out, S.Literal(sign * months, T.I32)], T.DATE)
            if micros or not months:
                out = S.Func("date_add", [out, S.Literal(sign * (micros // 86_400_000_000), T.I32)], T.DATE)
            return out
        if isinstance(other.dtype, T.TimestampType):
            total = sign * (micros + months * 2_592_000_000_000)  # months≈30d only if ts; Spark uses calendar — handled in add_months path
            out = other
            if months:
                out = S.Func("ts_add_months", [out, S.Literal(sign * months, T.I32)], T.TIMESTAMP)
            if micros or not months:
                out = S.BinaryOp("+", out, S.Literal(sign * micros, T.I64), T.TIMESTAMP)
            return out
        raise ResolutionError("interval arithmetic requires date/timestamp operand")
    if e.op in ("=", "!=", "<", "<=", ">", ">=", "<=>"):
        _coerce_pair(e)
        e.dtype = T.BOOL
        return e
    if e.op in ("+", "-", "*", "/", "%", "div"):
        # date +- int => date_add
        if e.op in ("+", "-") and isinstance(lt, T.DateType) and rt is not None and rt.is_integer:
            name = "date_add" if e.op == "+" else "date_sub"
            return S.Func(name, [e.left, e.right], T.DATE)
        if e.op == "-" and isinstance(lt, T.DateType) and isinstance(rt, T.DateType):
            return S.Func("datediff", [e.left, e.right], T.I32)
        if isinstance(lt, T.DecimalType) or isinstance(rt, T.DecimalType):
            return _type_decimal_arith(e)
        ct = T.common_type(lt, rt)
        if e.op == "/":
            # Spark: integer / integer -> double
            ct = T.F64 if not isinstance(ct, T.DecimalType) else ct
        if e.op == "div":
            ct = T.I64
        _cast_operands(e, ct if e.op != "/" else T.F64 if not isinstance(ct, T.DecimalType) else ct)
        e.dtype = ct
        return e
    raise ResolutionError(f"unknown binary op {e.op}")


def _type_decimal_arith(e: S.BinaryOp) -> S.BinaryOp:
    lt = _as_decimal(e.left)
    rt = _as_decimal(e.right)
    if e.op in ("+", "-"):
        scale = max(lt.scale, rt.scale)
        prec = max(lt.precision - lt.scale, rt.precision - rt.scale) + scale + 1
        e.dtype = T.DecimalType(min(38, prec), scale)
    elif e.op == "*":
        e.dtype = T.decimal_mul_type(lt, rt)
    elif e.op in ("/",):
        e.dtype = T.decimal_div_type(lt, rt)
    elif e.op in ("%",):
        scale = max(lt.scale, rt.scale)
        e.dtype = T.DecimalType(min(38, max(lt.precision, rt.precision)), scale)
    elif e.op == "div":
        e.dtype = T.I64
    # record operand decimal types via Cast insertion when int literals involved
    if not isinstance(e.left.dtype, T.DecimalType):
        e.left = S.Cast(e.left, lt, dtype=lt)
    if not isinstance(e.right.dtype, T.DecimalType):
        e.right = S.Cast(e.right, rt, dtype=rt)
    return e


def _as_decimal(x: S.Expr) -> T.DecimalType:
    t = x.dtype
    if isinstance(t, T.DecimalType):
        return t
    if t.is_integer:
        return T.DecimalType(19, 0)
    if t.is_float:
        # float wins: Spark converts decimal to double; approximate by scale-6 decimal
        return T.DecimalType(30, 6)
    raise ResolutionError(f"cannot use {t!r} in decimal arithmetic")


def _coerce_pair(e: S.BinaryOp):
    lt, rt = e.left.dtype, e.right.dtype
    if lt == rt:
        return
    # string literal vs date/timestamp: parse literal
    if isinstance(lt, T.DateType) and isinstance(rt, T.StringType) and isinstance(e.right, S.Literal):
        from ..sql.parser import _parse_date
        e.right = S.Literal(_parse_date(e.right.value), T.DATE)
        return
    if isinstance(rt, T.DateType) and isinstance(lt, T.StringType) and isinstance(e.left, S.Literal):
        from ..sql.parser import _parse_date
        e.left = S.Literal(_parse_date(e.left.value), T.DATE)
        return
    ct = T.common_type(lt, rt)
    _cast_operands(e, ct)


def _cast_operands(e: S.BinaryOp, ct: T.DataType):
    if e.left.dtype != ct:
        e.left = S.Cast(e.left, ct, dtype=ct)
    if e.right.dtype != ct:
        e.right = S.Cast(e.right, ct, dtype=ct)


def _coerce_to_bool(e: S.Expr) -> S.Expr:
    if e is None:
        return None
    if isinstance(e.dtype, T.BooleanType) or e.dtype is None:
        return e
    return S.Cast(e, T.BOOL, dtype=T.BOOL)


def _infer_literal_type(v) -> T.DataType:
    if v is None:
        return T.NULL
    if isinstance(v, bool):
        return T.BOOL
    if isinstance(v, int):
        return T.I32 if -(2 ** 31) <= v < 2 ** 31 else T.I64
    if isinstance(v, float):
        return T.F64
    if isinstance(v, str):
        return T.STRING
    return T.NULL


def _infer_pytype(values) -> T.DataType:
    for v in values:
        if v is None:
            continue
        return _infer_literal_type(v)
    return T.NULL


def _expr_name(e: S.Expr, i: int) -> str:
    if isinstance(e, S.Alias):
        return e.name
    if isinstance(e, (S.BoundRef, S.Col)):
        return e.name
    if isinstance(e, S.Cast):
        return _expr_name(e.child, i)
    if isinstance(e, S.AggFunc):
        return f"{e.name}({', '.join(_expr_name(a, i) for a in e.args) or '1'})"
    if isinstance(e, S.Func):
        return f"{e.name}({', '.join(_expr_name(a, i) for a in e.args)})"
    if isinstance(e, S.Literal):
        return str(e.value)
    return f"col{i}"


def _normalize_agg_name(name: str) -> str:
    name = name.lower()
    return {"mean": "avg", "first_value": "first", "last_value": "last",
            "some": "any", "bool_or": "any", "every": "bool_and",
            "array_agg": "collect_list", "stddev": "stddev_samp",
            "std": "stddev_samp", "percentile_cont": "percentile",
            "variance": "var_samp", "approx_percentile": "percentile_approx"}.get(name, name)


def _expr_equal_unbound(a: S.Expr, b: S.Expr) -> bool:
    """Structural equality between two *unbound* expressions (pre-resolution),
    used to match projection exprs against group-by exprs."""
    if isinstance(a, S.Alias):
        return _expr_equal_unbound(a.child, b)
    if isinstance(b, S.Alias):
        return _expr_equal_unbound(a, b.child)
    if type(a) is not type(b):
        return False
    if isinstance(a, S.Col):
        return a.name.lower() == b.name.lower() and (
            a.qualifier is None or b.qualifier is None
            or a.qualifier.lower() == b.qualifier.lower())
    if isinstance(a, S.Literal):
        return a.value == b.value
    if isinstance(a, S.BinaryOp) and a.op != b.op:
        return False
    if isinstance(a, S.UnaryOp) and a.op != b.op:
        return False
    if isinstance(a, S.Func) and a.name.lower() != b.name.lower():
        return False
    ca, cb = a.children(), b.children()
    if len(ca) != len(cb):
        return False
    return all(_expr_equal_unbound(x, y) for x, y in zip(ca, cb))


def _scope_fields(p: S.Plan) -> List[Field]:
    """Per-column qualifiers, preserved through nested join trees so that
    `n1.n_nationkey` resolves inside a 6-way comma join (q7/q8/q21)."""
    ovr = p.__dict__.get("_scope_fields_override") if hasattr(p, "__dict__") else None
    if ovr is not None:
        return ovr
    if isinstance(p, S.Join):
        if p.how in ("semi", "anti"):
            return _scope_fields(p.left)
        if p.how in ("rightsemi", "rightanti"):
            return _scope_fields(p.right)
        lf = _scope_fields(p.left)
        rf = _scope_fields(p.right)
        if p.using:
            used = {c.lower() for c in p.using}
            rf = [f for f in rf if f.name.lower() not in used]
        return lf + rf
    if isinstance(p, S.SubqueryAlias):
        return [Field(n, t, p.alias) for n, t in p.schema]
    if isinstance(p, (S.Filter, S.Limit, S.Sort, S.Distinct, S.Sample)):
        inner = _scope_fields(p.input)
        if len(inner) == len(p.schema):
            return inner
    if isinstance(p, S.Generate):
        inner = _scope_fields(p.input)
        gen = [Field(n, t, p.view_alias)
               for n, t in p.schema[len(p.input.schema):]]
        if len(inner) == len(p.input.schema):
            return inner + gen
        return [Field(n, t, None) for n, t in p.schema[:len(p.input.schema)]] + gen
    q = _plan_qualifier(p)
    return [Field(n, t, q) for n, t in p.schema]


def _plan_qualifier(p: S.Plan) -> Optional[str]:
    if isinstance(p, S.SubqueryAlias):
        return p.alias
    if isinstance(p, S.Read):
        return p.table.split(".")[-1]
    if isinstance(p, (S.Filter, S.Limit, S.Sort, S.Distinct, S.Sample)):
        return _plan_qualifier(p.input)
    return None


def _refs_table(p, name: str) -> bool:
    """Does the (unresolved) subtree read table `name`? (recursive-CTE detection)"""
    lname = name.lower()
    if isinstance(p, S.Read) and p.table.lower() == lname:
        return True
    for c in p.children():
        if c is not None and _refs_table(c, lname):
            return True
    # subqueries inside expressions
    for attr in ("exprs", "condition", "on"):
        v = getattr(p, attr, None)
        items = v if isinstance(v, list) else ([v] if v is not None else [])
        for e in items:
            if isinstance(e, S.Expr) and _expr_refs_table(e, lname):
                return True
    return False


def _expr_refs_table(e, name: str) -> bool:
    if isinstance(e, (S.ScalarSubquery, S.InSubquery, S.Exists)):
        if e.plan is not None and _refs_table(e.plan, name):
            return True
    for c in e.children():
        if isinstance(c, S.Expr) and _expr_refs_table(c, name):
            return True
    return False
