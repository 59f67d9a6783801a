"""RoaringBitmapArray (64-bit) + Z85 codecs for Delta deletion vectors.

(ref: crates/sail-delta-lake/src/deletion_vector/ — roaring bitmap +
z85 encoding.) Serialization follows the Delta protocol's portable
layout: an int64 count of 32-bit roaring bitmaps, each as
<int32 key><standard 32-bit roaring serialization>; 32-bit bitmaps use
the CRoaring portable format (cookie 12346 without run containers or
12347 with, array/bitmap/run containers, offset headers). The decoder
accepts all three container kinds; the encoder emits array containers
for sparse chunks and bitmap containers for dense ones.
"""
from __future__ import annotations

import struct
from typing import List, Sequence

import numpy as np

_Z85_CHARS = ("0123456789abcdefghijklmnopqrstuvwxyz"
              "ABCDEFGHIJKLMNOPQRSTUVWXYZ.-:+=^!/*?&<>()[]{}@%$#")
_Z85_DEC = {c: i for i, c in enumerate(_Z85_CHARS)}

SERIAL_COOKIE_NO_RUN = 12346
SERIAL_COOKIE = 12347
NO_OFFSET_THRESHOLD = 4


# ===========================================================================
# Z85 (ZeroMQ base-85; 4 bytes <-> 5 chars)
# ===========================================================================
def z85_encode(data: bytes) -> str:
    if len(data) % 4:
        raise ValueError("z85 requires length % 4 == 0")
    out = []
    for i in range(0, len(data), 4):
        v = struct.unpack(">I", data[i:i + 4])[0]
        block = []
        for _ in range(5):
            block.append(_Z85_CHARS[v % 85])
            v //= 85
        out.extend(reversed(block))
    return "".join(out)


def z85_decode(text: str) -> bytes:
    if len(text) % 5:
        raise ValueError("z85 requires length % 5 == 0")
    out = bytearray()
    for i in range(0, len(text), 5):
        v = 0
        for c in text[i:i + 5]:
            v = v * 85 + _Z85_DEC[c]
        out.extend(struct.pack(">I", v))
    return bytes(out)


# ===========================================================================
# 32-bit roaring
# ===========================================================================
def _serialize32(values: np.ndarray) -> bytes:
    """values: sorted uint32 array."""
    keys = (values >> 16).astype(np.uint32)
    lows = (values & 0xFFFF).astype(np.uint16)
    uk, starts = np.unique(keys, return_index=True)
    bounds = list(starts) + [len(values)]
    n = len(uk)
    out = bytearray()
    out += struct.pack("<iI", SERIAL_COOKIE_NO_RUN, n)
    containers = []
    for i in range(n):
        vals = lows[bounds[i]:bounds[i + 1]]
        card = len(vals)
        out += struct.pack("<HH", int(uk[i]), card - 1)
        if card <= 4096:
            containers.append(vals.tobytes())
        else:
            bits = np.zeros(65536 // 8, dtype=np.uint8)
            np.bitwise_or.at(bits, vals.astype(np.int64) // 8,
                             (1 << (vals.astype(np.int64) % 8)).astype(np.uint8))
            containers.append(bits.tobytes())
    # offset header (always present with the no-run cookie)
    pos = len(out) + 4 * n
    for c in containers:
        out += struct.pack("<I", pos)
        pos += len(c)
    for c in containers:
        out += c
    return bytes(out)


def _deserialize32(buf: memoryview, off: int) -> tuple:
    """Returns (uint32 ndarray, bytes consumed is not tracked — reads are
    offset-based via the header)."""
    (cookie,) = struct.unpack_from("<i", buf, off)
    run_flags = None
    if (cookie & 0xFFFF) == SERIAL_COOKIE:
        n = (cookie >> 16) + 1
        p = off + 4
        nb = (n + 7) // 8
        run_flags = np.unpackbits(
            np.frombuffer(buf, np.uint8, nb, p), bitorder="little")[:n]
        p += nb
        has_offsets = n >= NO_OFFSET_THRESHOLD
    elif cookie == SERIAL_COOKIE_NO_RUN:
        (n,) = struct.unpack_from("<I", buf, off + 4)
        p = off + 8
        has_offsets = True
    else:
        raise ValueError(f"bad roaring cookie {cookie}")
    keys = np.empty(n, np.uint32)
    cards = np.empty(n, np.int64)
    for i in range(n):
        k, c = struct.unpack_from("<HH", buf, p)
        keys[i] = k
        cards[i] = c + 1
        p += 4
    if has_offsets:
        p += 4 * n  # we read containers sequentially; offsets unused
    parts = []
    for i in range(n):
        is_run = run_flags is not None and run_flags[i]
        if is_run:
            (n_runs,) = struct.unpack_from("<H", buf, p)
            p += 2
            runs = np.frombuffer(buf, np.uint16, 2 * n_runs, p
                                 ).astype(np.uint32).reshape(n_runs, 2)
            p += 4 * n_runs
            vals = np.concatenate([np.arange(s, s + ln + 1, dtype=np.uint32)
                                   for s, ln in runs]) if n_runs else \
                np.empty(0, np.uint32)
        elif cards[i] <= 4096:
            vals = np.frombuffer(buf, np.uint16, cards[i], p).astype(np.uint32)
            p += 2 * cards[i]
        else:
            bits = np.frombuffer(buf, np.uint8, 8192, p)
            p += 8192
            vals = np.nonzero(np.unpackbits(bits, bitorder="little"))[0] \
                .astype(np.uint32)
        parts.append(vals + (np.uint32(keys[i]) << np.uint32(16)))
    return (np.concatenate(parts) if parts else np.empty(0, np.uint32)), p - off


# ===========================================================================
# 64-bit roaring bitmap array
# ===========================================================================
def roaring64_serialize(positions: Sequence[int]) -> bytes:
    vals = np.asarray(sorted(set(int(v) for v in positions)), dtype=np.uint64)
    highs = (vals >> np.uint64(32)).astype(np.uint32)
    lows = (vals & np.uint64(0xFFFFFFFF)).astype(np.uint32)
    uh, starts = np.unique(highs, return_index=True)
    bounds = list(starts) + [len(vals)]
    out = bytearray(struct.pack("<q", len(uh)))
    for i in range(len(uh)):
        out += struct.pack("<I", int(uh[i]))
        out += _serialize32(lows[bounds[i]:bounds[i + 1]])
    return bytes(out)


def roaring64_deserialize(data: bytes) -> np.ndarray:
    buf = memoryview(data)
    (n,) = struct.unpack_from("<q", buf, 0)
    p = 8
    parts: List[np.ndarray] = []
    for _ in range(n):
        (key,) = struct.unpack_from("<I", buf, p)
        p += 4
        vals, used = _deserialize32(buf, p)
        p += used
        parts.append(vals.astype(np.uint64) | (np.uint64(key) << np.uint64(32)))
    return (np.concatenate(parts) if parts
            else np.empty(0, np.uint64)).astype(np.int64)
