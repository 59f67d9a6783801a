"""Plan codec round-trip (plan/codec.py): serialize resolved+optimized
plans to JSON and back; decoded plans must execute to identical results
(the reference ships plans driver->worker through an analogous codec,
ref: crates/sail-execution)."""
import pytest

import sail_amd
from sail_amd.plan import spec as S
from sail_amd.plan.codec import (CodecError, plan_from_json, plan_to_json,
                                 type_from_obj, type_to_obj)


@pytest.fixture()
def s():
    ctx = sail_amd.SessionContext(device="cpu")
    ctx.create_dataframe({"k": ["a", "b", "a", "c"], "v": [1, 2, 3, 4],
                          "w": [1.5, 2.5, 3.5, 4.5]}, name="t1")
    ctx.create_dataframe({"k": ["a", "b"], "g": ["x", "y"]}, name="t2")
    return ctx


QUERIES = [
    "SELECT k, sum(v) s FROM t1 GROUP BY k HAVING sum(v) > 1 ORDER BY k",
    "SELECT t1.k, g, v*2 FROM t1 JOIN t2 ON t1.k = t2.k "
    "WHERE v BETWEEN 1 AND 5",
    "SELECT k, row_number() OVER (PARTITION BY k ORDER BY v) FROM t1",
    "SELECT CASE WHEN v > 1 THEN 'big' ELSE 'small' END, "
    "coalesce(NULL, k) FROM t1",
    "SELECT * FROM t1 WHERE k IN (SELECT k FROM t2 WHERE g = 'x')",
    "SELECT transform(array(1,2,3), x -> x + v) FROM t1",
    "SELECT date '2024-01-01' + make_dt_interval(1), "
    "CAST('1.50' AS decimal(5,2)), X'DEAD' FROM t1",
    "SELECT k, count(*) FROM t1 GROUP BY ROLLUP(k)",
    "SELECT named_struct('a', v, 'b', k), map('m', v) FROM t1",
]


@pytest.mark.parametrize("q", QUERIES)
def test_roundtrip_executes_identically(s, q):
    plan = s.plan_sql(q)
    back = plan_from_json(plan_to_json(plan))
    r1 = [tuple(c.to_pylist()) for c in s.execute_plan(plan).columns]
    r2 = [tuple(c.to_pylist()) for c in s.execute_plan(back).columns]
    assert r1 == r2


def test_type_codec_nested():
    from sail_amd.engine import types as T

    for t in (T.I64, T.STRING, T.BINARY, T.DecimalType(12, 3),
              T.GeometryType(3857), T.TIME,
              T.ArrayType(T.ArrayType(T.F64)),
              T.MapType(T.STRING, T.I64),
              T.StructType((T.StructField("a", T.I32),
                            T.StructField("b", T.ArrayType(T.STRING))))):
        assert type_from_obj(type_to_obj(t)) == t


def test_chunksource_not_serializable(s):
    from sail_amd.engine.chunk import Chunk

    node = S.ChunkSource(chunk=None, schema=None)
    with pytest.raises(CodecError):
        plan_to_json(node)


def test_binary_hex_literal(s):
    assert s.sql("SELECT X'DEAD'").collect() == [(b"\xde\xad",)]
    assert s.sql("SELECT hex(X'cafe')").collect() == [("CAFE",)]
