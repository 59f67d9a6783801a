"""Minimal Thrift compact-protocol reader for Parquet page headers.

Parquet page headers (PageHeader / DataPageHeader / DictionaryPageHeader,
parquet.thrift) are serialized with the Thrift *compact* protocol and sit
in front of every page in a column chunk. The reference reads them through
arrow-rs; here they are parsed directly so the GPU decoder
(datasource/gpu_parquet.py) can slice page regions without pyarrow.
ref: crates/sail-data-source/src/formats/parquet (reader surface).

Only what headers need is implemented: varint/zigzag ints, booleans,
binary, nested structs, and skipping of unknown fields (statistics).
"""
from __future__ import annotations

from typing import Dict, Tuple

# compact-protocol wire types
CT_STOP = 0x00
CT_TRUE = 0x01
CT_FALSE = 0x02
CT_BYTE = 0x03
CT_I16 = 0x04
CT_I32 = 0x05
CT_I64 = 0x06
CT_DOUBLE = 0x07
CT_BINARY = 0x08
CT_LIST = 0x09
CT_SET = 0x0A
CT_MAP = 0x0B
CT_STRUCT = 0x0C


def _varint(buf: bytes, pos: int) -> Tuple[int, int]:
    out = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        out |= (b & 0x7F) << shift
        if not (b & 0x80):
            return out, pos
        shift += 7


def _zigzag(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def _skip(buf: bytes, pos: int, wtype: int) -> int:
    if wtype in (CT_TRUE, CT_FALSE):
        return pos
    if wtype == CT_BYTE:
        return pos + 1
    if wtype in (CT_I16, CT_I32, CT_I64):
        _, pos = _varint(buf, pos)
        return pos
    if wtype == CT_DOUBLE:
        return pos + 8
    if wtype == CT_BINARY:
        n, pos = _varint(buf, pos)
        return pos + n
    if wtype in (CT_LIST, CT_SET):
        head = buf[pos]
        pos += 1
        n = head >> 4
        et = head & 0x0F
        if n == 15:
            n, pos = _varint(buf, pos)
        for _ in range(n):
            pos = _skip(buf, pos, et)
        return pos
    if wtype == CT_MAP:
        n, pos = _varint(buf, pos)
        if n:
            kv = buf[pos]
            pos += 1
            for _ in range(n):
                pos = _skip(buf, pos, kv >> 4)
                pos = _skip(buf, pos, kv & 0x0F)
        return pos
    if wtype == CT_STRUCT:
        _, pos = read_struct(buf, pos, parse=False)
        return pos
    raise ValueError(f"thrift compact: unknown wire type {wtype}")


def read_struct(buf: bytes, pos: int, parse: bool = True) -> Tuple[Dict[int, object], int]:
    """Parse one struct; returns ({field_id: value}, end_pos). Nested structs
    become dicts; bool fields True/False; ints decoded (zigzag); binary as
    bytes. With parse=False only advances pos."""
    out: Dict[int, object] = {}
    fid = 0
    while True:
        head = buf[pos]
        pos += 1
        if head == CT_STOP:
            return out, pos
        delta = head >> 4
        wtype = head & 0x0F
        if delta:
            fid += delta
        else:  # long-form field id: zigzag varint
            raw, pos = _varint(buf, pos)
            fid = _zigzag(raw)
        if not parse:
            pos = _skip(buf, pos, wtype)
            continue
        if wtype == CT_TRUE:
            out[fid] = True
        elif wtype == CT_FALSE:
            out[fid] = False
        elif wtype == CT_BYTE:
            out[fid] = buf[pos]
            pos += 1
        elif wtype in (CT_I16, CT_I32, CT_I64):
            raw, pos = _varint(buf, pos)
            out[fid] = _zigzag(raw)
        elif wtype == CT_DOUBLE:
            import struct as _s

            out[fid] = _s.unpack_from("<d", buf, pos)[0]
            pos += 8
        elif wtype == CT_BINARY:
            n, pos = _varint(buf, pos)
            out[fid] = bytes(buf[pos:pos + n])
            pos += n
        elif wtype == CT_STRUCT:
            out[fid], pos = read_struct(buf, pos)
        else:
            pos = _skip(buf, pos, wtype)
    raise AssertionError  # unreachable


# PageHeader field ids (parquet.thrift)
PAGE_TYPE = 1            # 0=DATA_PAGE 2=DICTIONARY_PAGE 3=DATA_PAGE_V2
UNCOMPRESSED_SIZE = 2
COMPRESSED_SIZE = 3
DATA_PAGE_HEADER = 5
DICT_PAGE_HEADER = 7
DATA_PAGE_HEADER_V2 = 8
# DataPageHeader
DPH_NUM_VALUES = 1
DPH_ENCODING = 2
DPH_DEF_ENCODING = 3
DPH_REP_ENCODING = 4
# DictionaryPageHeader
DICT_NUM_VALUES = 1
DICT_ENCODING = 2


def parse_page_header(buf: bytes, pos: int) -> Tuple[Dict[int, object], int]:
    """Parse a PageHeader at pos; returns (fields, first_byte_after_header)."""
    return read_struct(buf, pos)
