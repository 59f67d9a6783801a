"""Minimal Avro object-container-file codec (pure Python, stdlib only).

Iceberg's manifest lists and manifest files are Avro container files
(ref: crates/sail-iceberg/src/spec/ — manifest/manifest-list handling).
The image has no avro package, but the format is small: this module
implements the binary encoding (zigzag varints, length-prefixed bytes,
block-coded arrays/maps, union index tags) and the container framing
(magic "Obj\\x01", metadata map with the writer schema JSON, 16-byte sync
marker, deflate or null block codec) — enough to both read and write
Iceberg metadata, and to read generic Avro files.

Decoding is driven by the *writer schema embedded in the file*, so files
written by other engines (extra fields, different orders) decode
correctly into plain Python dicts/lists.
"""
from __future__ import annotations

import io
import json
import os
import struct
import zlib
from typing import Any, Dict, List, Optional, Tuple

MAGIC = b"Obj\x01"


# ===========================================================================
# binary encoding primitives
# ===========================================================================
def _read_long(buf: io.BytesIO) -> int:
    shift, acc = 0, 0
    while True:
        b = buf.read(1)
        if not b:
            raise EOFError("truncated varint")
        v = b[0]
        acc |= (v & 0x7F) << shift
        if not (v & 0x80):
            break
        shift += 7
    return (acc >> 1) ^ -(acc & 1)  # zigzag


def _write_long(out: bytearray, n: int):
    n = (n << 1) ^ (n >> 63)  # zigzag (python ints: arithmetic shift ok)
    if n < 0:
        n &= (1 << 64) - 1
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            break


def _read_bytes(buf: io.BytesIO) -> bytes:
    n = _read_long(buf)
    return buf.read(n)


def _write_bytes(out: bytearray, b: bytes):
    _write_long(out, len(b))
    out.extend(b)


# ===========================================================================
# schema-driven decode
# ===========================================================================
def _decode(schema, buf: io.BytesIO, names: Dict[str, Any]):
    if isinstance(schema, str):
        t = schema
        if t in names:
            return _decode(names[t], buf, names)
        if t == "null":
            return None
        if t == "boolean":
            return buf.read(1) != b"\x00"
        if t in ("int", "long"):
            return _read_long(buf)
        if t == "float":
            return struct.unpack("<f", buf.read(4))[0]
        if t == "double":
            return struct.unpack("<d", buf.read(8))[0]
        if t == "bytes":
            return _read_bytes(buf)
        if t == "string":
            return _read_bytes(buf).decode("utf-8")
        raise ValueError(f"unknown avro type {t!r}")
    if isinstance(schema, list):  # union
        idx = _read_long(buf)
        return _decode(schema[idx], buf, names)
    t = schema["type"]
    if t == "record":
        _register(schema, names)
        return {f["name"]: _decode(f["type"], buf, names)
                for f in schema["fields"]}
    if t == "array":
        out = []
        while True:
            n = _read_long(buf)
            if n == 0:
                break
            if n < 0:
                n = -n
                _read_long(buf)  # byte size, unused
            for _ in range(n):
                out.append(_decode(schema["items"], buf, names))
        return out
    if t == "map":
        out = {}
        while True:
            n = _read_long(buf)
            if n == 0:
                break
            if n < 0:
                n = -n
                _read_long(buf)
            for _ in range(n):
                k = _read_bytes(buf).decode("utf-8")
                out[k] = _decode(schema["values"], buf, names)
        return out
    if t == "fixed":
        _register(schema, names)
        return buf.read(schema["size"])
    if t == "enum":
        _register(schema, names)
        return schema["symbols"][_read_long(buf)]
    # logical types / wrapped primitives: {"type": "long", ...}
    return _decode(t, buf, names)


def _register(schema, names):
    nm = schema.get("name")
    if nm:
        ns = schema.get("namespace")
        names[nm] = schema
        if ns:
            names[f"{ns}.{nm}"] = schema


def _collect_names(schema, names):
    if isinstance(schema, dict):
        if schema.get("type") in ("record", "fixed", "enum"):
            _register(schema, names)
        for f in schema.get("fields", []) or []:
            _collect_names(f.get("type"), names)
        for k in ("items", "values"):
            if k in schema:
                _collect_names(schema[k], names)
    elif isinstance(schema, list):
        for s in schema:
            _collect_names(s, names)


# ===========================================================================
# schema-driven encode
# ===========================================================================
def _encode(schema, value, out: bytearray, names: Dict[str, Any]):
    if isinstance(schema, str):
        t = schema
        if t in names:
            return _encode(names[t], value, out, names)
        if t == "null":
            return
        if t == "boolean":
            out.append(1 if value else 0)
            return
        if t in ("int", "long"):
            _write_long(out, int(value))
            return
        if t == "float":
            out.extend(struct.pack("<f", float(value)))
            return
        if t == "double":
            out.extend(struct.pack("<d", float(value)))
            return
        if t == "bytes":
            _write_bytes(out, bytes(value))
            return
        if t == "string":
            _write_bytes(out, str(value).encode("utf-8"))
            return
        raise ValueError(f"unknown avro type {t!r}")
    if isinstance(schema, list):  # union: pick first matching branch
        idx = _union_index(schema, value)
        _write_long(out, idx)
        _encode(schema[idx], value, out, names)
        return
    t = schema["type"]
    if t == "record":
        _register(schema, names)
        for f in schema["fields"]:
            if f["name"] not in value and "default" in f:
                _encode(f["type"], f["default"], out, names)
            else:
                _encode(f["type"], value.get(f["name"]), out, names)
        return
    if t == "array":
        if value:
            _write_long(out, len(value))
            for v in value:
                _encode(schema["items"], v, out, names)
        _write_long(out, 0)
        return
    if t == "map":
        if value:
            _write_long(out, len(value))
            for k, v in value.items():
                _write_bytes(out, str(k).encode("utf-8"))
                _encode(schema["values"], v, out, names)
        _write_long(out, 0)
        return
    if t == "fixed":
        _register(schema, names)
        out.extend(bytes(value))
        return
    if t == "enum":
        _register(schema, names)
        _write_long(out, schema["symbols"].index(value))
        return
    _encode(t, value, out, names)


def _union_index(union, value) -> int:
    def kind(s):
        return s if isinstance(s, str) else (
            s["type"] if isinstance(s, dict) else None)

    if value is None:
        for i, s in enumerate(union):
            if kind(s) == "null":
                return i
        raise ValueError("null not allowed by union")
    for i, s in enumerate(union):
        if kind(s) != "null":
            return i
    raise ValueError("no non-null branch in union")


# ===========================================================================
# container files
# ===========================================================================
def read_container(path: str) -> Tuple[Any, List[Any], Dict[str, bytes]]:
    """Returns (writer_schema, records, metadata)."""
    with open(path, "rb") as f:
        data = f.read()
    buf = io.BytesIO(data)
    if buf.read(4) != MAGIC:
        raise ValueError(f"{path}: not an Avro container file")
    meta: Dict[str, bytes] = {}
    while True:
        n = _read_long(buf)
        if n == 0:
            break
        if n < 0:
            n = -n
            _read_long(buf)
        for _ in range(n):
            k = _read_bytes(buf).decode("utf-8")
            meta[k] = _read_bytes(buf)
    schema = json.loads(meta["avro.schema"].decode("utf-8"))
    codec = meta.get("avro.codec", b"null").decode("utf-8")
    sync = buf.read(16)
    names: Dict[str, Any] = {}
    _collect_names(schema, names)
    records: List[Any] = []
    while buf.tell() < len(data):
        count = _read_long(buf)
        size = _read_long(buf)
        block = buf.read(size)
        if codec == "deflate":
            block = zlib.decompress(block, -15)
        elif codec != "null":
            raise ValueError(f"unsupported avro codec {codec!r}")
        bb = io.BytesIO(block)
        for _ in range(count):
            records.append(_decode(schema, bb, names))
        if buf.read(16) != sync:
            raise ValueError(f"{path}: bad sync marker")
    return schema, records, meta


def write_container(path: str, schema, records: List[Any],
                    metadata: Optional[Dict[str, bytes]] = None,
                    codec: str = "deflate"):
    names: Dict[str, Any] = {}
    _collect_names(schema, names)
    body = bytearray()
    for r in records:
        _encode(schema, r, body, names)
    block = bytes(body)
    if codec == "deflate":
        c = zlib.compressobj(level=6, wbits=-15)
        block = c.compress(block) + c.flush()
    out = bytearray(MAGIC)
    meta = {"avro.schema": json.dumps(schema).encode("utf-8"),
            "avro.codec": codec.encode("utf-8")}
    meta.update(metadata or {})
    _write_long(out, len(meta))
    for k, v in meta.items():
        _write_bytes(out, k.encode("utf-8"))
        _write_bytes(out, v)
    _write_long(out, 0)
    sync = os.urandom(16)
    out.extend(sync)
    _write_long(out, len(records))
    _write_long(out, len(block))
    out.extend(block)
    out.extend(sync)
    tmp = path + ".tmp"
    with open(tmp, "wb") as f:
        f.write(bytes(out))
    os.replace(tmp, path)
