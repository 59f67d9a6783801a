"""Logical plan + expression IR.

Protocol-independent IR produced by both the SQL analyzer and the Spark
Connect proto converter, mirroring the role of the reference's spec layer
(ref: crates/sail-common/src/spec/plan.rs:34, expression.rs:13) with the
QueryNode/Expr vocabulary trimmed to what the engine executes.

Nodes carry *no* resolution state; the resolver (plan/resolver.py) produces a
typed bound tree.
"""
from __future__ import annotations

import itertools
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

from ..engine import types as T

_ids = itertools.count()


# ===========================================================================
# Expressions
# ===========================================================================

class Expr:
    """Base expression. After resolution, `dtype` is set."""

    dtype: Optional[T.DataType] = None

    def children(self) -> List["Expr"]:
        return []

    def with_children(self, ch: List["Expr"]) -> "Expr":
        if not ch:
            return self
        raise NotImplementedError(type(self))

    def walk(self):
        yield self
        for c in self.children():
            yield from c.walk()


@dataclass
class Literal(Expr):
    value: object
    dtype: Optional[T.DataType] = None

    def __repr__(self):
        return f"lit({self.value!r})"


@dataclass
class Col(Expr):
    """Unresolved attribute, possibly qualified ("t.a" -> qualifier="t")."""

    name: str
    qualifier: Optional[str] = None
    dtype: Optional[T.DataType] = None

    def __repr__(self):
        return f"col({self.qualifier + '.' if self.qualifier else ''}{self.name})"


@dataclass
class BoundRef(Expr):
    """Resolved reference to an input column by ordinal."""

    index: int
    name: str
    dtype: Optional[T.DataType] = None

    def __repr__(self):
        return f"#{self.index}:{self.name}"


@dataclass
class Alias(Expr):
    child: Expr
    name: str
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return Alias(ch[0], self.name, self.dtype)


@dataclass
class BinaryOp(Expr):
    """op in {+,-,*,/,%,=,!=,<,<=,>,>=,and,or,||}"""

    op: str
    left: Expr
    right: Expr
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.left, self.right]

    def with_children(self, ch):
        return BinaryOp(self.op, ch[0], ch[1], self.dtype)

    def __repr__(self):
        return f"({self.left!r} {self.op} {self.right!r})"


@dataclass
class UnaryOp(Expr):
    """op in {not, neg, isnull, isnotnull}"""

    op: str
    child: Expr
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return UnaryOp(self.op, ch[0], self.dtype)


@dataclass
class Cast(Expr):
    child: Expr
    to: T.DataType = None
    try_: bool = False
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return Cast(ch[0], self.to, self.try_, self.dtype)


@dataclass
class Func(Expr):
    """Scalar function call by Spark name (registry in functions/)."""

    name: str
    args: List[Expr] = field(default_factory=list)
    dtype: Optional[T.DataType] = None

    def children(self):
        return list(self.args)

    def with_children(self, ch):
        return Func(self.name, ch, self.dtype)

    def __repr__(self):
        return f"{self.name}({', '.join(map(repr, self.args))})"


@dataclass
class AggFunc(Expr):
    """Aggregate function: sum/avg/count/min/max/count_distinct/..."""

    name: str
    args: List[Expr] = field(default_factory=list)
    distinct: bool = False
    dtype: Optional[T.DataType] = None
    filter: Optional[Expr] = None

    def children(self):
        return list(self.args) + ([self.filter] if self.filter is not None else [])

    def with_children(self, ch):
        nargs = len(self.args)
        return AggFunc(self.name, ch[:nargs], self.distinct, self.dtype,
                       ch[nargs] if self.filter is not None else None)

    def __repr__(self):
        d = "distinct " if self.distinct else ""
        return f"{self.name}({d}{', '.join(map(repr, self.args))})"


@dataclass
class WindowExpr(Expr):
    func: Expr = None  # AggFunc or rank-like Func
    partition_by: List[Expr] = field(default_factory=list)
    order_by: List["SortKey"] = field(default_factory=list)
    frame: Optional[Tuple[str, object, object]] = None  # (rows|range, lo, hi)
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.func] + self.partition_by + [k.child for k in self.order_by]


@dataclass
class CaseWhen(Expr):
    branches: List[Tuple[Expr, Expr]] = field(default_factory=list)
    else_: Optional[Expr] = None
    dtype: Optional[T.DataType] = None

    def children(self):
        out = []
        for c, v in self.branches:
            out += [c, v]
        if self.else_ is not None:
            out.append(self.else_)
        return out

    def with_children(self, ch):
        n = len(self.branches)
        branches = [(ch[2 * i], ch[2 * i + 1]) for i in range(n)]
        els = ch[2 * n] if self.else_ is not None else None
        return CaseWhen(branches, els, self.dtype)


@dataclass
class InList(Expr):
    child: Expr = None
    values: List[Expr] = field(default_factory=list)
    negated: bool = False
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child] + self.values

    def with_children(self, ch):
        return InList(ch[0], ch[1:], self.negated, self.dtype)


@dataclass
class Between(Expr):
    child: Expr = None
    low: Expr = None
    high: Expr = None
    negated: bool = False
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child, self.low, self.high]

    def with_children(self, ch):
        return Between(ch[0], ch[1], ch[2], self.negated, self.dtype)


@dataclass
class Like(Expr):
    child: Expr = None
    pattern: Expr = None  # usually Literal
    negated: bool = False
    case_insensitive: bool = False
    is_regex: bool = False  # RLIKE
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child, self.pattern]

    def with_children(self, ch):
        return Like(ch[0], ch[1], self.negated, self.case_insensitive, self.is_regex, self.dtype)


@dataclass
class Lambda(Expr):
    """Higher-order-function lambda `x -> body` / `(x, i) -> body`
    (ref: Spark lambda functions; sail-plan resolver/expression/lambda.rs).
    After resolution, `body` references params as BoundRef 0..k-1 into the
    flattened-element chunk (outer columns follow at k+i)."""

    params: List[str] = field(default_factory=list)
    body: Expr = None
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.body]

    def with_children(self, ch):
        return Lambda(self.params, ch[0], self.dtype)

    def __repr__(self):
        return f"({', '.join(self.params)}) -> {self.body!r}"


@dataclass
class Star(Expr):
    qualifier: Optional[str] = None


@dataclass
class SortKey(Expr):
    child: Expr = None
    ascending: bool = True
    nulls_first: Optional[bool] = None  # None = Spark default (first if asc)

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return SortKey(ch[0], self.ascending, self.nulls_first)


@dataclass
class ScalarSubquery(Expr):
    plan: "Plan" = None
    dtype: Optional[T.DataType] = None


@dataclass
class InSubquery(Expr):
    child: Expr = None
    plan: "Plan" = None
    negated: bool = False
    dtype: Optional[T.DataType] = None

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return InSubquery(ch[0], self.plan, self.negated, self.dtype)


@dataclass
class Exists(Expr):
    plan: "Plan" = None
    negated: bool = False
    dtype: Optional[T.DataType] = None


# ===========================================================================
# Plans
# ===========================================================================

class Plan:
    """Base logical plan node. After resolution, `schema` is set to a list of
    (name, dtype) pairs."""

    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self) -> List["Plan"]:
        return []

    def field_names(self) -> List[str]:
        return [n for n, _ in (self.schema or [])]


@dataclass
class Read(Plan):
    """Read a named table (catalog lookup) — ref: spec::ReadType::NamedTable
    (crates/sail-common/src/spec/plan.rs:639)."""

    table: str
    schema: Optional[List[Tuple[str, T.DataType]]] = None
    options: Dict[str, str] = field(default_factory=dict)


@dataclass
class DataSourceRead(Plan):
    """read.format(...).load(paths) — ref: spec::ReadType::DataSource."""

    format: str
    paths: List[str] = field(default_factory=list)
    options: Dict[str, str] = field(default_factory=dict)
    user_schema: Optional[List[Tuple[str, T.DataType]]] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class LocalRelation(Plan):
    """Literal table (VALUES / createDataFrame)."""

    data: Dict[str, list] = field(default_factory=dict)
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class TableFuncRead(Plan):
    """FROM my_udtf(args...) — user-defined table function
    (ref: Spark UDTF; sail-python-udf table functions)."""

    name: str = ""
    args: List[Expr] = field(default_factory=list)
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class ChunkSource(Plan):
    """Pre-materialized chunk spliced into a plan (streaming incremental
    aggregation substitutes the Aggregate subtree with its merged state)."""

    chunk: object = None  # engine.chunk.Chunk
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class Project(Plan):
    input: Plan = None
    exprs: List[Expr] = field(default_factory=list)
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Filter(Plan):
    input: Plan = None
    condition: Expr = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Join(Plan):
    left: Plan = None
    right: Plan = None
    how: str = "inner"  # inner|left|right|full|semi|anti|cross|existence
    on: Optional[Expr] = None  # condition; None for cross
    using: Optional[List[str]] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.left, self.right]


@dataclass
class Aggregate(Plan):
    input: Plan = None
    group_by: List[Expr] = field(default_factory=list)
    aggs: List[Expr] = field(default_factory=list)  # Alias(AggFunc) or exprs over groups
    grouping_sets: Optional[List[List[int]]] = None  # rollup/cube expansions
    having: Optional[Expr] = None  # bound by the aggregate resolver
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Sort(Plan):
    input: Plan = None
    keys: List[SortKey] = field(default_factory=list)
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Limit(Plan):
    input: Plan = None
    n: Optional[int] = None
    offset: int = 0
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class RecursionRef(Plan):
    """Reference to the working set inside a recursive CTE body
    (ref: sail-plan resolver/query/recursion.rs; sail-logical-plan
    RecursiveCTE nodes)."""

    name: str = ""
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class RecursiveCte(Plan):
    """WITH RECURSIVE name AS (anchor UNION [ALL] recursive): iterate the
    recursive term to a fixpoint."""

    name: str = ""
    anchor: Plan = None
    recursive: Plan = None
    is_all: bool = True
    max_iter: int = 100
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.anchor, self.recursive]


@dataclass
class Pivot(Plan):
    """t PIVOT (agg(v) FOR k IN (v1, v2, ...)) — resolved into a grouped
    aggregate with one filtered aggregate per pivot value (ref: Spark pivot;
    sail-plan resolver/query/pivoting.rs role)."""

    input: Plan = None
    agg: Expr = None            # AggFunc over the value column (unresolved)
    pivot: Expr = None          # the pivot column expression
    values: List[Expr] = field(default_factory=list)  # literals
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Unpivot(Plan):
    """t UNPIVOT (val FOR name IN (c1, c2, ...)) — resolved into a UNION ALL
    of per-column projections."""

    input: Plan = None
    value_name: str = "value"
    name_name: str = "name"
    columns: List[str] = field(default_factory=list)
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Generate(Plan):
    """Generator (explode/posexplode): one output row per array element,
    parent columns repeated — ref: Spark Generate; spec generator functions
    (crates/sail-plan/src/function/generator.rs)."""

    input: Plan = None
    gen: Expr = None          # the array-valued expression
    outer: bool = False       # explode_outer: keep empty/null arrays as null row
    position: bool = False    # posexplode: emit 0-based pos column
    mode: str = ""            # "" | "inline" (array<struct> field expansion)
    #: LATERAL VIEW form: generated-column names + view alias (unresolved)
    aliases: Optional[List[str]] = None
    view_alias: Optional[str] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Sample(Plan):
    """TABLESAMPLE (n PERCENT | n ROWS) [REPEATABLE (seed)] — ref: Spark
    sample grammar; spec::Sample (crates/sail-common/src/spec/plan.rs)."""

    input: Plan = None
    fraction: Optional[float] = None
    rows: Optional[int] = None
    seed: Optional[int] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Distinct(Plan):
    input: Plan = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class SetOp(Plan):
    """union|intersect|except, with is_all / by_name flags
    (ref: crates/sail-common/src/spec/plan.rs:723)."""

    op: str = "union"
    left: Plan = None
    right: Plan = None
    is_all: bool = False
    by_name: bool = False
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.left, self.right]


@dataclass
class SubqueryAlias(Plan):
    input: Plan = None
    alias: str = ""
    column_aliases: Optional[List[str]] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class WindowPlan(Plan):
    input: Plan = None
    window_exprs: List[Expr] = field(default_factory=list)  # Alias(WindowExpr)
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class Range(Plan):
    """spark.range(start, end, step) — ref: spec::QueryNode::Range."""

    start: int = 0
    end: int = 0
    step: int = 1
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class WithCte(Plan):
    """WITH name AS (...) ... — inlined by the resolver."""

    ctes: List[Tuple[str, Plan]] = field(default_factory=list)
    input: Plan = None
    recursive: bool = False
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


# --- commands (DDL / writes) ----------------------------------------------

@dataclass
class Command(Plan):
    pass


@dataclass
class CreateView(Command):
    name: str = ""
    input: Plan = None
    replace: bool = False
    temporary: bool = True
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class CreateTable(Command):
    name: str = ""
    columns: List[Tuple[str, T.DataType]] = field(default_factory=list)
    input: Optional[Plan] = None  # CTAS
    format: Optional[str] = None
    location: Optional[str] = None
    replace: bool = False
    if_not_exists: bool = False
    options: Dict[str, str] = field(default_factory=dict)
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class DropTable(Command):
    name: str = ""
    if_exists: bool = False
    is_view: bool = False
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class InsertInto(Command):
    table: str = ""
    input: Plan = None
    overwrite: bool = False
    columns: Optional[List[str]] = None  # INSERT INTO t (a, b) ...
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class Write(Command):
    """df.write.format(...).save(path) — ref: spec::CommandNode::Write."""

    input: Plan = None
    format: str = "parquet"
    path: Optional[str] = None
    table: Optional[str] = None
    mode: str = "error"  # append|overwrite|error|ignore
    partition_by: List[str] = field(default_factory=list)
    options: Dict[str, str] = field(default_factory=dict)
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class Explain(Command):
    input: Plan = None
    mode: str = "simple"
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class SetConfig(Command):
    key: str = ""
    value: Optional[str] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class VacuumTable(Command):
    name: str = ""
    retention_hours: Optional[float] = None
    dry_run: bool = False


@dataclass
class DescribeHistory(Command):
    name: str = ""


@dataclass
class AlterTable(Command):
    """ALTER TABLE: add/drop/rename columns, rename table (in-memory
    catalog + delta metadata — ref: spec CommandNode AlterTable subset)."""

    name: str = ""
    action: str = ""            # add_columns | drop_column | rename_column | rename_table
    columns: List[Tuple[str, T.DataType]] = field(default_factory=list)
    column: str = ""
    new_name: str = ""


@dataclass
class ShowFunctions(Command):
    pattern: Optional[str] = None


@dataclass
class ShowDatabases(Command):
    pass


@dataclass
class CacheTable(Command):
    """CACHE TABLE name [AS query]: materialize a view/query as a table
    (ref: spec CommandNode cache/uncache)."""

    name: str = ""
    input: Optional[Plan] = None


@dataclass
class UncacheTable(Command):
    name: str = ""


@dataclass
class AnalyzeTable(Command):
    """ANALYZE TABLE ... COMPUTE STATISTICS [FOR COLUMNS ...]: populate the
    planner's column statistics (ref: spec AnalyzeTable)."""

    name: str = ""
    columns: Optional[List[str]] = None


@dataclass
class ShowTables(Command):
    pattern: Optional[str] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class DescribeTable(Command):
    name: str = ""
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class DescribeQuery(Command):
    input: Plan = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.input]


@dataclass
class ShowColumns(Command):
    name: str = ""


@dataclass
class ShowCreateTable(Command):
    name: str = ""


@dataclass
class ShowViews(Command):
    pattern: Optional[str] = None


@dataclass
class ShowPartitions(Command):
    name: str = ""


@dataclass
class ShowTblProperties(Command):
    name: str = ""


@dataclass
class ShowCatalogs(Command):
    pass


@dataclass
class UseDatabase(Command):
    name: str = ""


@dataclass
class CreateDatabase(Command):
    name: str = ""
    if_not_exists: bool = False
    comment: str = ""


@dataclass
class DropDatabase(Command):
    name: str = ""
    if_exists: bool = False
    cascade: bool = False


@dataclass
class RefreshTable(Command):
    name: str = ""


@dataclass
class TruncateTable(Command):
    name: str = ""


@dataclass
class CommentOn(Command):
    kind: str = "table"  # table | column
    name: str = ""
    comment: Optional[str] = None


# ---------------------------------------------------------------------------

def plan_tree_string(plan: Plan, indent: int = 0) -> str:
    pad = "  " * indent
    name = type(plan).__name__
    extra = ""
    if isinstance(plan, Read):
        extra = f" table={plan.table}"
    elif isinstance(plan, Filter):
        extra = f" cond={plan.condition!r}"
    elif isinstance(plan, Project):
        extra = f" exprs={plan.exprs!r}"
    elif isinstance(plan, Join):
        extra = f" how={plan.how} on={plan.on!r}"
    elif isinstance(plan, Aggregate):
        extra = f" keys={plan.group_by!r} aggs={plan.aggs!r}"
    elif isinstance(plan, Sort):
        extra = f" keys={plan.keys!r}"
    elif isinstance(plan, Limit):
        extra = f" n={plan.n}"
    elif isinstance(plan, SubqueryAlias):
        extra = f" alias={plan.alias}"
    out = f"{pad}{name}{extra}\n"
    for c in plan.children():
        out += plan_tree_string(c, indent + 1)
    return out


@dataclass
class OuterRef(Expr):
    """Reference to a column of an *outer* query from inside a subquery.
    Produced by the resolver; eliminated by the decorrelator."""

    index: int = 0
    name: str = ""
    dtype: Optional[T.DataType] = None

    def __repr__(self):
        return f"outer#{self.index}:{self.name}"


@dataclass
class MergeAction:
    """One WHEN clause of MERGE (ref: spec::CommandNode::MergeInto,
    crates/sail-common/src/spec/plan.rs MergeInto)."""

    kind: str  # "update" | "delete" | "insert" | "insert_star" | "update_star"
    condition: Optional[Expr] = None
    assignments: List[Tuple[str, Expr]] = field(default_factory=list)
    insert_columns: Optional[List[str]] = None
    insert_values: Optional[List[Expr]] = None


@dataclass
class MergeInto(Command):
    target: str = ""
    target_alias: Optional[str] = None
    source: Plan = None
    source_alias: Optional[str] = None
    on: Expr = None
    matched: List[MergeAction] = field(default_factory=list)
    not_matched: List[MergeAction] = field(default_factory=list)
    not_matched_by_source: List[MergeAction] = field(default_factory=list)
    schema: Optional[List[Tuple[str, T.DataType]]] = None

    def children(self):
        return [self.source] if self.source is not None else []


@dataclass
class UpdateTable(Command):
    table: str = ""
    assignments: List[Tuple[str, Expr]] = field(default_factory=list)
    condition: Optional[Expr] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None


@dataclass
class DeleteFrom(Command):
    table: str = ""
    condition: Optional[Expr] = None
    schema: Optional[List[Tuple[str, T.DataType]]] = None
