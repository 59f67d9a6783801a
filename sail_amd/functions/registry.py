"""Spark function registry.

Mirrors the reference's function name registry
(ref: crates/sail-plan/src/function/scalar/*, aggregate.rs, window.rs) —
names and semantics from Spark, implementations our own. Each scalar function
declares a return-type rule; implementations live in engine/eval.py (torch
path, runs on CPU and on ROCm) and, for the hot subset, as ExprVM opcodes in
ops/csrc/exprvm.hip.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional

from ..engine import types as T

# ---------------------------------------------------------------------------
# Aggregate functions (ref: sail-plan function/aggregate.rs ~85 names)
# ---------------------------------------------------------------------------

AGG_FUNCTIONS = {
    "hll_sketch_agg", "hll_union_agg", "theta_sketch_agg", "theta_union_agg",
    "tuple_sketch_agg_double", "tuple_sketch_agg_integer",
    "tuple_union_agg_double", "tuple_union_agg_integer",
    "tuple_intersection_agg_double", "tuple_intersection_agg_integer",
    "sum", "avg", "mean", "count", "min", "max", "first", "first_value",
    "last", "last_value", "stddev", "stddev_samp", "stddev_pop", "variance",
    "var_samp", "var_pop", "count_if", "any", "some", "bool_or", "every",
    "bool_and", "collect_list", "array_agg", "collect_set", "approx_count_distinct",
    "corr", "covar_samp", "covar_pop", "skewness", "kurtosis", "median",
    "percentile", "percentile_approx", "approx_percentile", "mode", "product",
    "sum_distinct", "bit_and", "bit_or", "bit_xor", "max_by", "min_by",
    "any_value", "try_sum", "try_avg", "regr_count", "regr_avgx", "regr_avgy",
    "regr_slope", "regr_intercept", "regr_r2", "regr_sxx", "regr_syy", "regr_sxy",
    "grouping", "grouping_id", "histogram_numeric", "string_agg", "listagg",
    "std", "percentile_cont", "percentile_disc", "vector_sum", "vector_avg",
}

WINDOW_FUNCTIONS = {
    "row_number", "rank", "dense_rank", "percent_rank", "cume_dist", "ntile",
    "lag", "lead", "nth_value",
}


def agg_return_type(name: str, arg_types: List[T.DataType], distinct: bool = False) -> T.DataType:
    from ..engine.aggregates import UDAFS

    if name not in UDAFS and "_agg" in name:
        from ..engine import functions_impl  # noqa: F401  (sketch UDAFs)
    if name in UDAFS:
        return UDAFS[name][1]
    a = arg_types[0] if arg_types else T.NULL
    if name in ("count", "count_if", "approx_count_distinct", "regr_count"):
        return T.I64
    if name in ("sum", "try_sum", "sum_distinct"):
        if isinstance(a, T.DecimalType):
            return T.DecimalType(min(38, a.precision + 10), a.scale)
        if a.is_integer:
            return T.I64
        return T.F64
    if name in ("avg", "mean", "try_avg"):
        if isinstance(a, T.DecimalType):
            return T.DecimalType(min(38, a.precision + 4), min(a.scale + 4, 10))
        return T.F64
    if name in ("min", "max", "first", "first_value", "last", "last_value",
                "any_value", "mode", "median", "max_by", "min_by",
                "percentile_disc"):
        return a
    if name in ("vector_sum", "vector_avg"):
        return T.ArrayType(T.F64)
    if name == "histogram_numeric":
        return T.ArrayType(T.StructType((T.StructField("x", T.F64),
                                         T.StructField("y", T.F64))))
    if name in ("stddev", "stddev_samp", "stddev_pop", "variance", "var_samp",
                "std", "percentile_cont",
                "var_pop", "corr", "covar_samp", "covar_pop", "skewness",
                "kurtosis", "percentile", "percentile_approx", "approx_percentile",
                "product") or name.startswith("regr_"):
        return T.F64
    if name in ("any", "some", "bool_or", "every", "bool_and"):
        return T.BOOL
    if name in ("collect_list", "array_agg", "collect_set"):
        return T.ArrayType(a)
    if name in ("bit_and", "bit_or", "bit_xor"):
        return a if a.is_integer else T.I64
    if name in ("grouping", "grouping_id"):
        return T.I32 if name == "grouping" else T.I64
    if name in ("string_agg", "listagg"):
        return T.STRING
    raise KeyError(f"unknown aggregate {name}")


# ---------------------------------------------------------------------------
# Scalar function type inference
# ---------------------------------------------------------------------------

def _same(args):
    return args[0]


def _f64(args):
    return T.F64


def _i32(args):
    return T.I32


def _i64(args):
    return T.I64


def _bool(args):
    return T.BOOL


def _string(args):
    return T.STRING


def _date(args):
    return T.DATE


def _ts(args):
    return T.TIMESTAMP


def _numeric_common(args):
    t = args[0]
    for a in args[1:]:
        t = T.common_type(t, a)
    return t


# name -> return-type rule.  Grouped like the reference's scalar modules
# (ref: crates/sail-function/src/scalar/).
SCALAR_RETURN: Dict[str, Callable[[List[T.DataType]], T.DataType]] = {}


def _reg(names, rule):
    for n in names.split():
        SCALAR_RETURN[n] = rule


# math (ref: sail-function/src/scalar/math)
_reg("abs", _same)
_reg("ceil ceiling floor", lambda a: T.I64 if not isinstance(a[0], T.DecimalType) else T.DecimalType(a[0].precision - a[0].scale + 1, 0))
_reg("round bround", lambda a: a[0])
_reg("sqrt cbrt exp expm1 ln log log10 log2 log1p sin cos tan asin acos atan "
     "sinh cosh tanh asinh acosh atanh atan2 degrees radians pi e rand randn "
     "power pow hypot", _f64)
_reg("sign signum", _f64)
_reg("mod pmod", _numeric_common)
_reg("greatest least", _numeric_common)
_reg("factorial", _i64)
_reg("bin hex unhex conv", _string)
_reg("bitwise_not shiftleft shiftright shiftrightunsigned bit_count", lambda a: a[0] if a and a[0].is_integer else T.I32)
_reg("positive negative", _same)
_reg("width_bucket", _i64)
_reg("try_add try_subtract try_multiply try_divide", _numeric_common)
_reg("rint", _f64)
_reg("log", _f64)

# string (ref: sail-function/src/scalar/string)
_reg("concat_ws upper ucase lower lcase trim ltrim rtrim btrim initcap "
     "reverse repeat lpad rpad substring substr left right replace translate "
     "regexp_replace regexp_extract regexp_extract_all split_part soundex "
     "chr char space format_string printf format_number substring_index "
     "overlay sentences elt base64 unbase64 decode encode to_char", _string)
_reg("length len char_length character_length octet_length bit_length "
     "instr locate position levenshtein crc32 ascii find_in_set", _i32)
_reg("startswith endswith contains like ilike rlike regexp regexp_like", _bool)
_reg("split", lambda a: T.ArrayType(T.STRING))
_reg("md5 sha sha1 sha2 uuid current_version", _string)
_reg("hash", _i32)
_reg("xxhash64", _i64)

# datetime (ref: sail-function/src/scalar/datetime)
_reg("year month day dayofmonth dayofweek dayofyear weekday weekofyear quarter "
     "hour minute second", _i32)
_reg("date_part_doy date_part_dow", _i32)
_reg("datediff date_diff", _i32)
_reg("date_add dateadd date_sub add_months last_day next_day to_date trunc", _date)
_reg("date_trunc to_timestamp timestamp_seconds timestamp_millis timestamp_micros "
     "from_unixtime from_utc_timestamp to_utc_timestamp make_timestamp", _ts)
_reg("current_date curdate", _date)
_reg("current_timestamp now localtimestamp", _ts)
_reg("unix_timestamp to_unix_timestamp unix_seconds unix_millis unix_micros "
     "unix_date", _i64)
_reg("months_between", _f64)
_reg("date_format", _string)
_reg("make_date", _date)
_reg("extract", _i32)

# conditional / misc (ref: sail-function/src/scalar/misc, predicate)
_reg("coalesce nvl ifnull", _numeric_common)
_reg("nvl2", lambda a: T.common_type(a[1], a[2]))
_reg("nullif", _same)
_reg("if iff", lambda a: T.common_type(a[1], a[2]))
_reg("isnull isnotnull isnan", _bool)
_reg("nanvl", _f64)
_reg("assert_true raise_error", lambda a: T.NULL)
_reg("monotonically_increasing_id spark_partition_id input_file_block_start "
     "input_file_block_length", _i64)
_reg("input_file_name current_user current_database current_catalog version", _string)
_reg("typeof", _string)
_reg("get_field", lambda a: T.NULL)  # resolved structurally
_reg("struct named_struct", lambda a: T.StructType(tuple(T.StructField(f"col{i+1}", t) for i, t in enumerate(a))))

# array/collection (subset; ref: sail-function/src/scalar/array, collection)
_reg("array", lambda a: T.ArrayType(a[0] if a else T.NULL))
_reg("array_contains", _bool)
_reg("size cardinality array_size", _i32)
_reg("array_max array_min", lambda a: a[0].element if isinstance(a[0], T.ArrayType) else T.NULL)
def _elem_rule(a):
    if isinstance(a[0], T.ArrayType):
        return a[0].element
    if isinstance(a[0], T.MapType):
        return a[0].value
    return T.NULL


_reg("element_at element_at_sql get try_element_at", _elem_rule)
_reg("map", lambda a: T.MapType(a[0] if a else T.STRING, a[1] if len(a) > 1 else T.STRING))
_reg("map_from_arrays", lambda a: T.MapType(a[0].element, a[1].element))
_reg("map_keys", lambda a: T.ArrayType(a[0].key) if isinstance(a[0], T.MapType) else T.NULL)
_reg("map_values", lambda a: T.ArrayType(a[0].value) if isinstance(a[0], T.MapType) else T.NULL)
_reg("map_contains_key", _bool)
_reg("struct named_struct get_field", lambda a: T.NULL)  # typed structurally in the resolver
_reg("get_json_object to_json schema_of_json json_tuple", _string)
_reg("json_array_length", _i32)
_reg("json_object_keys", lambda a: T.ArrayType(T.STRING))
_reg("from_csv", lambda a: T.NULL)  # typed structurally in the resolver
_reg("to_csv schema_of_csv to_xml schema_of_xml collation", _string)
_reg("collate", lambda a: a[0])
_reg("from_xml", lambda a: T.NULL)  # typed structurally in the resolver
_reg("parse_url try_parse_url url_encode url_decode", _string)
_reg("xpath", lambda a: T.ArrayType(T.STRING))
_reg("xpath_string", _string)
_reg("xpath_boolean", _bool)
_reg("xpath_int xpath_short", _i32)
_reg("xpath_long", _i64)
_reg("xpath_double xpath_float xpath_number", _f64)
_reg("parse_json try_parse_json schema_of_variant", _string)
_reg("variant_get try_variant_get", lambda a: T.STRING)  # refined at eval
                                                         # when type literal
                                                         # given
_reg("is_variant_null luhn_check is_valid_utf8 is_valid_variant", _bool)
_reg("cot csc sec", _f64)
_reg("current_timezone variant_to_json to_variant_object try_url_decode "
     "validate_utf8 try_validate_utf8 make_valid_utf8", _string)
_reg("binary", lambda a: T.BINARY)
_reg("random_poisson bitmap_bit_position bitmap_bucket_number deep_size", _i64)
_reg("crc32c", _i64)
_reg("array_insert", lambda a: a[0])
_reg("array_contains_all", _bool)
_reg("array_concat", lambda a: a[0])
_reg("concat", lambda a: a[0] if a and isinstance(a[0], T.ArrayType) else T.STRING)
_reg("map_concat", lambda a: a[0])
_reg("str_to_map", lambda a: T.MapType(T.STRING, T.STRING))
_reg("vector_norm vector_inner_product vector_l2_distance", _f64)
_reg("vector_normalize", lambda a: T.ArrayType(T.F64))
_reg("arrays_zip map_entries map_from_entries", lambda a: T.NULL)  # resolver-typed
_reg("strpos regexp_count regexp_instr", _i32)
_reg("quote mask regexp_substr dayname to_varchar to_char randstr "
     "current_schema user session_user", _string)
_reg("date_from_unix_date", lambda a: T.DATE)
_reg("timestampadd timestamp_add convert_timezone", lambda a: T.TIMESTAMP)
_reg("timestampdiff timestamp_diff uniform", _i64)
_reg("random", _f64)
_reg("to_number try_to_number", _f64)
_reg("to_binary try_to_binary", lambda a: T.BINARY)
_reg("getbit bit_get", _i32)
_reg("try_mod", lambda a: a[0])
_reg("equal_null", _bool)
_reg("nullifzero zeroifnull", lambda a: a[0])
_reg("from_json", lambda a: T.NULL)  # typed structurally in the resolver
_reg("date_format from_unixtime", _string)
_reg("to_timestamp try_to_timestamp", lambda a: T.TIMESTAMP)
_reg("datepart date_part", _i32)
_reg("window", lambda a: T.StructType((T.StructField("start", T.TIMESTAMP),
                                       T.StructField("end", T.TIMESTAMP))))
_reg("window_time", lambda a: T.TIMESTAMP)
_reg("session_window", lambda a: T.StructType((
    T.StructField("start", T.TIMESTAMP), T.StructField("end", T.TIMESTAMP))))
_reg("sort_array array_sort array_distinct array_remove array_compact flatten "
     "slice array_repeat shuffle", _same)
_reg("array_join", _string)
_reg("array_position", _i64)
_reg("arrays_overlap", _bool)
_reg("array_union array_intersect array_except", _same)
_reg("array_append array_prepend", _same)
_reg("array_repeat", lambda a: T.ArrayType(a[0]))
_reg("flatten", lambda a: a[0].element if isinstance(a[0], T.ArrayType) else T.NULL)
_reg("sequence", lambda a: T.ArrayType(a[0]))
_reg("explode explode_outer posexplode posexplode_outer",
     lambda a: a[0].element if isinstance(a[0], T.ArrayType) else T.NULL)
_reg("inline inline_outer",
     lambda a: a[0].element if isinstance(a[0], T.ArrayType) else T.NULL)
_reg("stack", lambda a: T.NULL)  # shape resolved by the generator binder

# type conversion helpers
_reg("double", _f64)
_reg("float", lambda a: T.F32)
_reg("int integer", _i32)
_reg("bigint long", _i64)
_reg("smallint", lambda a: T.I16)
_reg("tinyint", lambda a: T.I8)
_reg("boolean", _bool)
_reg("string", _string)
_reg("decimal", lambda a: T.DecimalType(10, 0))
_reg("date", _date)
_reg("timestamp", _ts)


def scalar_return_type(name: str, arg_types: List[T.DataType]) -> Optional[T.DataType]:
    rule = SCALAR_RETURN.get(name)
    if rule is None:
        return None
    return rule(arg_types)


# round-2 extension batch (impls in engine/functions_ext.py)
_reg("monthname", lambda a: T.STRING)
_reg("add_days add_years", lambda a: T.DATE)
_reg("div bitmap_count hll_sketch_estimate theta_sketch_estimate",
     lambda a: T.I64)
_reg("aes_encrypt try_aes_encrypt aes_decrypt try_aes_decrypt "
     "hll_union theta_union theta_intersection theta_difference",
     lambda a: T.BINARY)
_reg("cosine_similarity vector_cosine_similarity l1", lambda a: T.F64)
_reg("unbase64 unhex", lambda a: T.BINARY)  # Spark returns BINARY
_reg("crc32", lambda a: T.I64)  # Spark: BIGINT (values exceed int32)
_reg("try_divide", lambda a: T.F64 if not isinstance(a[0], T.DecimalType)
     else T.DecimalType(min(38, a[0].precision + 4), min(a[0].scale + 4, 10)))
_reg("regexp_extract_all", lambda a: T.ArrayType(T.STRING))
_reg("sentences", lambda a: T.ArrayType(T.ArrayType(T.STRING)))
_reg("time to_time try_to_time make_time time_trunc current_time "
     "time_from_micros time_from_millis time_from_seconds",
     lambda a: T.TIME)
_reg("time_to_micros time_to_millis time_to_seconds time_diff",
     lambda a: T.I64)
_reg("to_avro", lambda a: T.BINARY)
_reg("from_avro", lambda a: T.NULL)   # real type resolved at eval (schema arg)
_reg("schema_of_avro", lambda a: T.STRING)
_reg("to_timestamp_ltz to_timestamp_ntz make_timestamp_ltz make_timestamp_ntz "
     "try_make_timestamp try_make_timestamp_ltz try_make_timestamp_ntz "
     "time_bucket", _ts)
_reg("years", _i32)
_reg("tuple_union_double tuple_union_integer tuple_intersection_double "
     "tuple_intersection_integer tuple_difference_double "
     "tuple_difference_integer tuple_union_theta_double "
     "tuple_union_theta_integer tuple_intersection_theta_double "
     "tuple_intersection_theta_integer tuple_difference_theta_double "
     "tuple_difference_theta_integer tuple_sketch_theta_double "
     "tuple_sketch_theta_integer", lambda a: T.BINARY)
_reg("tuple_sketch_estimate_double tuple_sketch_estimate_integer "
     "tuple_sketch_summary_double", _f64)
_reg("tuple_sketch_summary_integer", _i64)
_reg("tuple_sketch_estimate", lambda a: T.F64)

_reg("to_protobuf", lambda a: T.BINARY)
_reg("from_protobuf", lambda a: T.NULL)  # struct type resolved at eval

# geo (ref: sail-plan/src/function/scalar/geo.rs; WKB + SRID model)
_reg("st_geomfromwkb", lambda a: T.GeometryType())
_reg("st_geogfromwkb", lambda a: T.GeographyType())
_reg("st_asbinary", lambda a: T.BINARY)
_reg("st_srid", lambda a: T.I32)
_reg("st_setsrid", lambda a: a[0])  # SRID literal applied at eval
