"""Numpy simulator of the parquet_decode.hip kernels.

Implements the exact kernel contracts (page-descriptor tables, dense
outputs) so the host orchestration in datasource/gpu_parquet.py can be
validated on CPU; the HIP kernels themselves are checked against pyarrow
on the GPU (tests/test_gpu.py)."""
from __future__ import annotations

import numpy as np
import torch


def _np(t):
    return t.cpu().numpy()


def _read_varint(buf, pos):
    out = 0
    shift = 0
    while True:
        b = int(buf[pos])
        pos += 1
        out |= (b & 0x7F) << shift
        if not (b & 0x80):
            return out, pos
        shift += 7


def _zigzag(v):
    return (v >> 1) ^ -(v & 1)


def _wrap64(v):
    v &= (1 << 64) - 1
    return v - (1 << 64) if v >= (1 << 63) else v


def _read_bits(buf, bit, bw):
    byte = bit >> 3
    sh = bit & 7
    need = (sh + bw + 7) >> 3
    v = 0
    for i in range(need):
        v |= int(buf[byte + i]) << (8 * i)
    v >>= sh
    if bw < 64:
        v &= (1 << bw) - 1
    return v


def pq_rle_decode(buf_t, pages_t, total):
    buf = _np(buf_t)
    pages = _np(pages_t).reshape(-1, 6)
    out = np.zeros(total, dtype=np.int32)
    for src_off, src_len, nvals, out_row, _aux, bw in pages:
        src = buf[src_off:src_off + src_len]
        pos = 0
        v = 0
        while v < nvals and pos < src_len:
            h, pos = _read_varint(src, pos)
            if h & 1:
                groups = h >> 1
                cnt = min(groups * 8, nvals - v)
                for i in range(cnt):
                    out[out_row + v + i] = _read_bits(src, (pos * 8) + i * bw, bw)
                pos += groups * bw
            else:
                cnt = min(h >> 1, nvals - v)
                nb = (bw + 7) // 8
                val = 0
                for i in range(nb):
                    val |= int(src[pos + i]) << (8 * i)
                pos += nb
                out[out_row + v:out_row + v + cnt] = val
            v += cnt
    return torch.from_numpy(out)


def pq_plain_copy(buf_t, pages_t, total, width):
    buf = _np(buf_t)
    out = np.zeros(total * width, dtype=np.uint8)
    for src_off, src_len, nvals, out_row, _aux, w in _np(pages_t).reshape(-1, 6):
        n = nvals * width
        out[out_row * width:out_row * width + n] = buf[src_off:src_off + n]
    return torch.from_numpy(out)


def pq_copy_bytes(buf_t, pages_t, out_t):
    buf = _np(buf_t)
    out = out_t.numpy()
    for src_off, src_len, _n, _r, aux, _w in _np(pages_t).reshape(-1, 6):
        out[aux:aux + src_len] = buf[src_off:src_off + src_len]


def pq_flba_i64(buf_t, pages_t, total, width):
    buf = _np(buf_t)
    out = np.zeros(total, dtype=np.int64)
    for src_off, _len, nvals, out_row, _aux, _w in _np(pages_t).reshape(-1, 6):
        b = buf[src_off:src_off + nvals * width].reshape(nvals, width)
        v = np.zeros(nvals, dtype=object)
        for i in range(width):
            v = (v * 256) + b[:, i].astype(object)
        sign = 1 << (8 * width - 1)
        v = np.where(np.array([x & sign for x in v], dtype=bool),
                     np.array([x - (1 << (8 * width)) for x in v], dtype=object), v)
        out[out_row:out_row + nvals] = v.astype(np.int64)
    return torch.from_numpy(out)


def pq_delta_decode(buf_t, pages_t, total):
    buf = _np(buf_t)
    pages = _np(pages_t).reshape(-1, 6)
    out = np.zeros(total, dtype=np.int64)
    data_end = np.zeros(len(pages), dtype=np.int64)
    for pi, (src_off, src_len, nvals, out_row, _aux, _w) in enumerate(pages):
        src = buf[src_off:src_off + src_len]
        pos = 0
        block_size, pos = _read_varint(src, pos)
        mbpb, pos = _read_varint(src, pos)
        tot, pos = _read_varint(src, pos)
        first_raw, pos = _read_varint(src, pos)
        out[out_row] = _wrap64(_zigzag(first_raw))
        vpm = block_size // mbpb
        n = min(tot, nvals)
        ndeltas = max(n - 1, 0)
        d = 0
        while d < ndeltas and pos < src_len:
            md_raw, pos = _read_varint(src, pos)
            min_delta = _zigzag(md_raw)
            bws = src[pos:pos + mbpb]
            pos += mbpb
            for k in range(mbpb):
                if d >= ndeltas:
                    break
                bw = int(bws[k])
                cnt = min(vpm, ndeltas - d)
                for i in range(cnt):
                    v = 0 if bw == 0 else _read_bits(src, pos * 8 + i * bw, bw)
                    out[out_row + 1 + d + i] = _wrap64(min_delta + v)
                d += cnt
                pos += vpm * bw // 8
        data_end[pi] = pos
    return torch.from_numpy(out), torch.from_numpy(data_end)


def pq_bytearray_walk(buf_t, pages_t, total):
    buf = _np(buf_t)
    lengths = np.zeros(total, dtype=np.int64)
    src_pos = np.zeros(total, dtype=np.int64)
    for src_off, src_len, nvals, out_row, _aux, _w in _np(pages_t).reshape(-1, 6):
        pos = 0
        for v in range(nvals):
            if pos + 4 > src_len:
                break
            ln = int.from_bytes(bytes(buf[src_off + pos:src_off + pos + 4]),
                                "little")
            lengths[out_row + v] = ln
            src_pos[out_row + v] = src_off + pos + 4
            pos += 4 + ln
    return torch.from_numpy(lengths), torch.from_numpy(src_pos)


def pq_gather_strings(buf_t, src_pos_t, lengths_t, out_offsets_t, total_bytes):
    buf = _np(buf_t)
    out = np.zeros(total_bytes, dtype=np.uint8)
    for sp, ln, oo in zip(_np(src_pos_t), _np(lengths_t), _np(out_offsets_t)):
        out[oo:oo + ln] = buf[sp:sp + ln]
    return torch.from_numpy(out)


def pq_segscan(data_t, pages_t):
    data = data_t.numpy()
    for _off, _len, nvals, out_row, aux, _w in _np(pages_t).reshape(-1, 6):
        seg = data[out_row:out_row + nvals]
        np.cumsum(seg, out=seg)
        seg += aux
