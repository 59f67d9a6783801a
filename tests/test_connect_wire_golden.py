"""Independent wire-encoding validation.

The in-repo Connect client and server share wire.py field constants, so a
shared field-number mistake would pass every conformance test. Here the
same Relation/Expression messages are built through google.protobuf with
descriptors declared from the PUBLIC spark/connect proto field numbers
(relations.proto / expressions.proto) and compared byte-for-byte against
our hand-rolled builders — an encoder our code shares nothing with.
"""
import pytest

from sail_amd.connect.client import E, R
from sail_amd.connect import wire as W


def _pool():
    from google.protobuf import descriptor_pb2, descriptor_pool

    fds = descriptor_pb2.FileDescriptorSet()
    fd = fds.file.add()
    fd.name = "spark/connect/mini.proto"
    fd.package = "spark.connect"
    fd.syntax = "proto3"

    def msg(name):
        m = fd.message_type.add()
        m.name = name
        return m

    def field(m, name, num, ftype, type_name=None, repeated=False):
        f = m.field.add()
        f.name = name
        f.number = num
        f.type = ftype
        f.label = 3 if repeated else 1
        if type_name:
            f.type_name = ".spark.connect." + type_name
        return f

    MSG, STR, I32, I64, BOOL = 11, 9, 5, 3, 8

    # Expression subset (expressions.proto)
    lit = msg("Literal")
    field(lit, "long", 7, I64)
    field(lit, "string", 13, STR)
    attr = msg("UnresolvedAttribute")
    field(attr, "unparsed_identifier", 1, STR)
    fn = msg("UnresolvedFunction")
    field(fn, "function_name", 1, STR)
    field(fn, "arguments", 2, MSG, "Expression", repeated=True)
    field(fn, "is_distinct", 3, BOOL)
    alias = msg("Alias")
    field(alias, "expr", 1, MSG, "Expression")
    field(alias, "name", 2, STR, repeated=True)
    expr = msg("Expression")
    field(expr, "literal", 1, MSG, "Literal")
    field(expr, "unresolved_attribute", 2, MSG, "UnresolvedAttribute")
    field(expr, "unresolved_function", 3, MSG, "UnresolvedFunction")
    field(expr, "alias", 6, MSG, "Alias")

    # Relation subset (relations.proto)
    named = msg("NamedTable")
    field(named, "unparsed_identifier", 1, STR)
    read = msg("Read")
    field(read, "named_table", 1, MSG, "NamedTable")
    project = msg("Project")
    field(project, "input", 1, MSG, "Relation")
    field(project, "expressions", 3, MSG, "Expression", repeated=True)
    filt = msg("Filter")
    field(filt, "input", 1, MSG, "Relation")
    field(filt, "condition", 2, MSG, "Expression")
    agg = msg("Aggregate")
    field(agg, "input", 1, MSG, "Relation")
    field(agg, "group_type", 2, 14, "Aggregate.GroupType")
    field(agg, "grouping_expressions", 3, MSG, "Expression", repeated=True)
    field(agg, "aggregate_expressions", 4, MSG, "Expression", repeated=True)
    en = agg.enum_type.add()
    en.name = "GroupType"
    for i, n in enumerate(["GROUP_TYPE_UNSPECIFIED", "GROUP_TYPE_GROUPBY",
                           "GROUP_TYPE_ROLLUP", "GROUP_TYPE_CUBE",
                           "GROUP_TYPE_PIVOT", "GROUP_TYPE_GROUPING_SETS"]):
        v = en.value.add()
        v.name = n
        v.number = i
    limit = msg("Limit")
    field(limit, "input", 1, MSG, "Relation")
    field(limit, "limit", 2, I32)
    rel = msg("Relation")
    field(rel, "read", 2, MSG, "Read")
    field(rel, "project", 3, MSG, "Project")
    field(rel, "filter", 4, MSG, "Filter")
    field(rel, "limit", 8, MSG, "Limit")
    field(rel, "aggregate", 9, MSG, "Aggregate")

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fd)
    return pool


def _cls(pool, name):
    from google.protobuf import message_factory

    return message_factory.GetMessageClass(
        pool.FindMessageTypeByName("spark.connect." + name))


def test_relation_bytes_match_protobuf_library():
    pool = _pool()
    Relation = _cls(pool, "Relation")

    # SELECT k, sum(v) AS sv FROM rt WHERE v > 2 GROUP BY k LIMIT 5 as a
    # relation tree, built by google.protobuf:
    m = Relation()
    lim = m.limit
    lim.limit = 5
    ag = lim.input.aggregate
    ag.group_type = 1
    flt = ag.input.filter
    flt.input.read.named_table.unparsed_identifier = "rt"
    cond = flt.condition.unresolved_function
    cond.function_name = ">"
    cond.arguments.add().unresolved_attribute.unparsed_identifier = "v"
    cond.arguments.add().literal.long = 2
    ag.grouping_expressions.add().unresolved_attribute.unparsed_identifier \
        = "k"
    al = ag.aggregate_expressions.add().alias
    al.name.append("sv")
    sm = al.expr.unresolved_function
    sm.function_name = "sum"
    sm.arguments.add().unresolved_attribute.unparsed_identifier = "v"
    want = m.SerializeToString()

    ours = R.limit(
        R.aggregate(
            R.filter(R.read_table("rt"),
                     E.fn(">", E.col("v"), E.lit_long(2))),
            group=[E.col("k")],
            aggs=[E.alias(E.fn("sum", E.col("v")), "sv")]),
        5)
    assert ours == want


def test_expression_bytes_match_protobuf_library():
    pool = _pool()
    Expression = _cls(pool, "Expression")

    e = Expression()
    f = e.unresolved_function
    f.function_name = "count"
    f.arguments.add().unresolved_attribute.unparsed_identifier = "x"
    f.is_distinct = True
    assert E.fn("count", E.col("x"), distinct=True) == e.SerializeToString()

    e2 = Expression()
    e2.literal.string = "hello"
    assert E.lit_str("hello") == e2.SerializeToString()
