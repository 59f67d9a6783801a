"""Device-side string assembly for synthetic data generation.

Builds Arrow-layout (offsets, bytes) string columns entirely with tensor ops,
so TPC-H text columns (comments, p_name, names) can be generated directly in
HBM without host round-trips. Works identically on CPU for tests.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..engine.column import StringColumn, _pack_strings


def pack_vocab(words: List[str], device) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    offs, byts = _pack_strings(words, device)
    lens = (offs[1:] - offs[:-1]).to(torch.int64)
    return offs, byts, lens


def assemble_words(word_ids: torch.Tensor, nwords: torch.Tensor,
                   vocab: Tuple[torch.Tensor, torch.Tensor, torch.Tensor],
                   sep: int = 32, row_chunk: int = 8_000_000) -> StringColumn:
    """Build one string per row by joining words from a vocabulary.

    word_ids: [n, kmax] int64 vocabulary indices
    nwords:   [n] number of words used per row (<= kmax)

    Processed in row chunks: the per-byte index temporaries are ~24 B per
    output byte, which at SF100 (16 GB of comment text) would otherwise need
    ~400 GB transient HBM.
    """
    n = word_ids.shape[0]
    if n <= row_chunk:
        return _assemble_words_chunk(word_ids, nwords, vocab, sep)
    parts = [
        _assemble_words_chunk(word_ids[i : i + row_chunk], nwords[i : i + row_chunk], vocab, sep)
        for i in range(0, n, row_chunk)
    ]
    dev = word_ids.device
    lens = torch.cat([p.offsets[1:] - p.offsets[:-1] for p in parts])
    offsets = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=offsets[1:])
    byts = torch.cat([p.bytes_ for p in parts])
    return StringColumn(offsets, byts)


def _assemble_words_chunk(word_ids: torch.Tensor, nwords: torch.Tensor,
                          vocab, sep: int) -> StringColumn:
    voffs, vbytes, vlens = vocab
    n, kmax = word_ids.shape
    dev = word_ids.device
    valid = torch.arange(kmax, device=dev).unsqueeze(0) < nwords.unsqueeze(1)  # [n,kmax]
    wlen = torch.where(valid, vlens[word_ids], torch.zeros_like(word_ids))
    # +1 separator after each word except the last of each row
    is_last = torch.arange(kmax, device=dev).unsqueeze(0) == (nwords - 1).unsqueeze(1)
    span = wlen + (valid & ~is_last).to(torch.int64)
    row_len = span.sum(dim=1)
    offsets = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(row_len, 0, out=offsets[1:])
    total = int(offsets[-1].item())
    # flatten valid words in row-major order
    keep = valid.reshape(-1)
    spans = span.reshape(-1)[keep]
    ids = word_ids.reshape(-1)[keep]
    del valid, wlen, is_last, span, keep
    # destination start of each word
    dstart = torch.cumsum(spans, 0)
    dstart = dstart - spans
    # per byte: which word, offset within word (int32 per-chunk indices)
    byte_word = torch.repeat_interleave(
        torch.arange(spans.shape[0], device=dev, dtype=torch.int64), spans)
    off_in = torch.arange(total, device=dev) - dstart.index_select(0, byte_word)
    wl = vlens[ids].index_select(0, byte_word)
    src = voffs[ids].index_select(0, byte_word) + torch.minimum(off_in, wl - 1).clamp_min(0)
    del dstart, byte_word
    out = torch.where(off_in < wl, vbytes.index_select(0, src),
                      torch.full((1,), sep, dtype=torch.uint8, device=dev))
    return StringColumn(offsets, out)


def keyed_names(prefix: str, keys: torch.Tensor, width: int = 9) -> StringColumn:
    """'Customer#000000001'-style fixed-width names, assembled on device."""
    dev = keys.device
    n = keys.shape[0]
    pre = torch.frombuffer(bytearray(prefix.encode()), dtype=torch.uint8).to(dev)
    plen = pre.shape[0]
    rowlen = plen + width
    offsets = torch.arange(0, (n + 1) * rowlen, rowlen, dtype=torch.int64, device=dev)
    out = torch.empty(n * rowlen, dtype=torch.uint8, device=dev)
    out.view(n, rowlen)[:, :plen] = pre.unsqueeze(0)
    k = keys.to(torch.int64)
    for i in range(width):
        digit = torch.div(k, 10 ** (width - 1 - i), rounding_mode="floor") % 10
        out.view(n, rowlen)[:, plen + i] = (digit + 48).to(torch.uint8)
    return StringColumn(offsets, out)
