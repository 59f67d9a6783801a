"""Minimal protobuf wire-format encoder/decoder.

The image has no protoc/grpcio-tools, so Spark Connect messages are read and
written directly at the wire level (varint tags, length-delimited fields).
Unknown fields are skipped on decode (forward compatibility), matching
protobuf semantics.
"""
from __future__ import annotations

from typing import Dict, Iterator, List, Tuple, Union

WIRE_VARINT = 0
WIRE_I64 = 1
WIRE_LEN = 2
WIRE_I32 = 5


def encode_varint(v: int) -> bytes:
    out = bytearray()
    if v < 0:
        v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7


def tag(field: int, wire: int) -> bytes:
    return encode_varint((field << 3) | wire)


def field_varint(field: int, v: int) -> bytes:
    return tag(field, WIRE_VARINT) + encode_varint(v)


def field_bytes(field: int, data: bytes) -> bytes:
    return tag(field, WIRE_LEN) + encode_varint(len(data)) + data


def field_string(field: int, s: str) -> bytes:
    return field_bytes(field, s.encode("utf-8"))


def field_message(field: int, payload: bytes) -> bytes:
    return field_bytes(field, payload)


def iter_fields(buf: bytes) -> Iterator[Tuple[int, int, Union[int, bytes]]]:
    """Yields (field_number, wire_type, value). LEN fields yield bytes."""
    pos = 0
    n = len(buf)
    while pos < n:
        key, pos = decode_varint(buf, pos)
        field = key >> 3
        wire = key & 7
        if wire == WIRE_VARINT:
            v, pos = decode_varint(buf, pos)
            yield field, wire, v
        elif wire == WIRE_LEN:
            ln, pos = decode_varint(buf, pos)
            yield field, wire, buf[pos : pos + ln]
            pos += ln
        elif wire == WIRE_I64:
            yield field, wire, int.from_bytes(buf[pos : pos + 8], "little")
            pos += 8
        elif wire == WIRE_I32:
            yield field, wire, int.from_bytes(buf[pos : pos + 4], "little")
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wire}")


def parse(buf: bytes) -> Dict[int, List[Union[int, bytes]]]:
    """All fields grouped by number (repeated-friendly)."""
    out: Dict[int, List[Union[int, bytes]]] = {}
    for f, _, v in iter_fields(buf):
        out.setdefault(f, []).append(v)
    return out


def first(fields: Dict[int, List], num: int, default=None):
    vals = fields.get(num)
    return vals[0] if vals else default


def first_str(fields: Dict[int, List], num: int, default: str = "") -> str:
    v = first(fields, num)
    return v.decode("utf-8") if isinstance(v, (bytes, bytearray)) else default


def first_varint(fields: Dict[int, List], num: int, default: int = 0) -> int:
    v = first(fields, num)
    return int(v) if isinstance(v, int) else default


def all_strs(fields, num) -> list:
    """All values of a repeated string field."""
    out = []
    for v in fields.get(num, []):
        out.append(v.decode("utf-8") if isinstance(v, bytes) else str(v))
    return out
