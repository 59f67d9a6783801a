"""Data type system.

Mirrors the Spark/Arrow type surface the reference exposes
(ref: crates/sail-common/src/spec/data_type.rs:105) but with a device-first
storage mapping: every fixed-width type is backed by a torch tensor dtype,
decimals are scaled int64, strings are Arrow-style (offsets, bytes) pairs.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional, Tuple

import torch


class DataType:
    """Base class for engine data types."""

    #: torch dtype used for physical storage (None for nested/string types)
    storage: Optional[torch.dtype] = None

    def __eq__(self, other):
        return type(self) is type(other)

    def __hash__(self):
        return hash(type(self))

    def __repr__(self):
        return type(self).__name__.replace("Type", "").lower()

    # -- classification helpers -------------------------------------------
    @property
    def is_numeric(self) -> bool:
        return isinstance(self, (IntegerLike, FloatLike, DecimalType))

    @property
    def is_integer(self) -> bool:
        return isinstance(self, IntegerLike)

    @property
    def is_float(self) -> bool:
        return isinstance(self, FloatLike)

    @property
    def is_string(self) -> bool:
        return isinstance(self, StringType)

    @property
    def is_temporal(self) -> bool:
        return isinstance(self, (DateType, TimestampType, TimeType))


class IntegerLike(DataType):
    bits: int = 64
    signed: bool = True


class FloatLike(DataType):
    bits: int = 64


class NullType(DataType):
    pass


class BooleanType(DataType):
    storage = torch.bool


class Int8Type(IntegerLike):
    storage = torch.int8
    bits = 8


class Int16Type(IntegerLike):
    storage = torch.int16
    bits = 16


class Int32Type(IntegerLike):
    storage = torch.int32
    bits = 32


class Int64Type(IntegerLike):
    storage = torch.int64
    bits = 64


class Float32Type(FloatLike):
    storage = torch.float32
    bits = 32


class Float64Type(FloatLike):
    storage = torch.float64
    bits = 64


class DateType(DataType):
    """Days since epoch (Arrow date32)."""

    storage = torch.int32


class TimeType(DataType):
    """Microseconds since midnight (Arrow time64[us]; Spark 4.1 TIME)."""

    storage = torch.int64


class TimestampType(DataType):
    """Microseconds since epoch (Arrow timestamp[us]). Session-timezone-naive
    storage; tz handling happens at the boundary like the reference
    (ref: crates/sail-common/src/spec/data_type.rs Timestamp*)."""

    storage = torch.int64


@dataclass(frozen=True, eq=True)
class DecimalType(DataType):
    """decimal(precision, scale) stored as scaled int64.

    Spark decimals up to precision 18 fit int64; wider precisions produced by
    aggregation keep int64 storage with int128 accumulation inside kernels
    (documented deviation: storage precision caps at 18 significant digits
    beyond the scale)."""

    precision: int = 18
    scale: int = 2

    @property
    def storage(self):  # type: ignore[override]
        return torch.int64

    def __repr__(self):
        return f"decimal({self.precision},{self.scale})"


class StringType(DataType):
    """UTF-8 string; physical layout is (offsets int64, bytes uint8),
    optionally dictionary-encoded (codes int32 + unique values)."""

    storage = None


class BinaryType(StringType):
    pass


@dataclass(frozen=True, eq=True)
class GeometryType(BinaryType):
    """2D spatial data, Cartesian coordinates; stored as WKB bytes with an
    SRID (Spark 4.1 GeometryType; ref: crates/sail-common/src/spec/
    data_type.rs:273-287 — geoarrow.wkb extension over Binary). Physical
    layout is the binary (offsets, bytes) pair."""

    srid: int = 4326

    def __repr__(self):
        return f"geometry({self.srid})"


@dataclass(frozen=True, eq=True)
class GeographyType(BinaryType):
    """2D spatial data, spherical (geodetic) coordinates; WKB + SRID
    (Spark 4.1 GeographyType, spherical edge interpolation)."""

    srid: int = 4326

    def __repr__(self):
        return f"geography({self.srid})"


@dataclass(frozen=True, eq=True)
class ArrayType(DataType):
    element: DataType = field(default_factory=Int64Type)

    def __repr__(self):
        return f"array<{self.element!r}>"


@dataclass(frozen=True, eq=True)
class StructField:
    name: str
    dtype: DataType
    nullable: bool = True


@dataclass(frozen=True, eq=True)
class StructType(DataType):
    fields: Tuple[StructField, ...] = ()

    def __repr__(self):
        inner = ",".join(f"{f.name}:{f.dtype!r}" for f in self.fields)
        return f"struct<{inner}>"


@dataclass(frozen=True, eq=True)
class MapType(DataType):
    key: DataType = field(default_factory=StringType)
    value: DataType = field(default_factory=StringType)

    def __repr__(self):
        return f"map<{self.key!r},{self.value!r}>"


# Singletons for the common types
NULL = NullType()
BOOL = BooleanType()
I8 = Int8Type()
I16 = Int16Type()
I32 = Int32Type()
I64 = Int64Type()
F32 = Float32Type()
F64 = Float64Type()
DATE = DateType()
TIME = TimeType()
TIMESTAMP = TimestampType()
STRING = StringType()
BINARY = BinaryType()

_BY_NAME = {
    "null": NULL, "void": NULL,
    "boolean": BOOL, "bool": BOOL,
    "tinyint": I8, "byte": I8,
    "smallint": I16, "short": I16,
    "int": I32, "integer": I32,
    "bigint": I64, "long": I64,
    "float": F32, "real": F32,
    "double": F64,
    "date": DATE,
    "timestamp": TIMESTAMP, "timestamp_ltz": TIMESTAMP, "timestamp_ntz": TIMESTAMP,
    "time": TIME,
    "string": STRING, "varchar": STRING, "char": STRING, "text": STRING,
    "binary": BINARY,
}


def type_name(t: DataType) -> str:
    """Canonical SQL name (inverse of type_from_name)."""
    if isinstance(t, DecimalType):
        return f"decimal({t.precision},{t.scale})"
    m = {NullType: "void", BooleanType: "boolean", Int8Type: "tinyint",
         Int16Type: "smallint", Int32Type: "int", Int64Type: "bigint",
         Float32Type: "float", Float64Type: "double", DateType: "date",
         TimestampType: "timestamp", TimeType: "time", StringType: "string",
         BinaryType: "binary"}
    if type(t) in m:
        return m[type(t)]
    return str(t)


def type_from_name(name: str) -> DataType:
    base = name.strip().lower()
    if base.startswith("geometry") or base.startswith("geography"):
        srid = int(base[base.find("(") + 1:base.find(")")]) \
            if "(" in base else 4326
        return GeometryType(srid) if base.startswith("geometry") \
            else GeographyType(srid)
    if base.startswith("decimal") or base.startswith("numeric"):
        inner = base[base.find("(") + 1 : base.find(")")] if "(" in base else "10,0"
        p, _, s = inner.partition(",")
        return DecimalType(int(p), int(s or 0))
    if "(" in base:  # varchar(n), char(n)
        base = base[: base.find("(")]
    if base in _BY_NAME:
        return _BY_NAME[base]
    raise ValueError(f"unknown type name: {name!r}")


# ---------------------------------------------------------------------------
# Type coercion (Spark-style numeric promotion)
# ---------------------------------------------------------------------------

_INT_ORDER = {Int8Type: 0, Int16Type: 1, Int32Type: 2, Int64Type: 3}


def common_type(a: DataType, b: DataType) -> DataType:
    """Least common type for binary arithmetic / comparison, Spark semantics
    (ref: crates/sail-plan/src/resolver/expression/cast.rs behavior)."""
    if a == b:
        return a
    if isinstance(a, NullType):
        return b
    if isinstance(b, NullType):
        return a
    if isinstance(a, FloatLike) or isinstance(b, FloatLike):
        if isinstance(a, Float64Type) or isinstance(b, Float64Type):
            return F64
        # float32 vs any int/decimal -> float64 like Spark's widening for safety
        if isinstance(a, Float32Type) and isinstance(b, Float32Type):
            return F32
        return F64
    if isinstance(a, DecimalType) or isinstance(b, DecimalType):
        # promote both sides to decimal
        da = a if isinstance(a, DecimalType) else _int_as_decimal(a, a)
        db = b if isinstance(b, DecimalType) else _int_as_decimal(b, b)
        scale = max(da.scale, db.scale)
        ip = max(da.precision - da.scale, db.precision - db.scale)
        return DecimalType(min(38, ip + scale), scale)
    if isinstance(a, IntegerLike) and isinstance(b, IntegerLike):
        return a if _INT_ORDER[type(a)] >= _INT_ORDER[type(b)] else b
    if isinstance(a, StringType) and b.is_numeric:
        return F64
    if isinstance(b, StringType) and a.is_numeric:
        return F64
    if isinstance(a, DateType) and isinstance(b, StringType):
        return DATE
    if isinstance(b, DateType) and isinstance(a, StringType):
        return DATE
    if isinstance(a, TimestampType) and isinstance(b, (DateType, StringType)):
        return TIMESTAMP
    if isinstance(b, TimestampType) and isinstance(a, (DateType, StringType)):
        return TIMESTAMP
    raise TypeError(f"no common type for {a!r} and {b!r}")


def _int_as_decimal(t: DataType, _orig) -> DecimalType:
    if isinstance(t, DecimalType):
        return t
    bits = getattr(t, "bits", 64)
    prec = {8: 3, 16: 5, 32: 10, 64: 19}.get(bits, 19)
    return DecimalType(min(prec, 38), 0)


def decimal_mul_type(a: DecimalType, b: DecimalType) -> DecimalType:
    return DecimalType(min(38, a.precision + b.precision + 1), a.scale + b.scale)


def decimal_div_type(a: DecimalType, b: DecimalType) -> DecimalType:
    # Spark: scale = max(6, s1 + p2 + 1)
    scale = max(6, a.scale + b.precision + 1)
    prec = a.precision - a.scale + b.scale + scale
    # clamp for int64 storage pragmatics
    scale = min(scale, 12)
    return DecimalType(min(38, prec), scale)
