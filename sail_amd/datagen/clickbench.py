"""Synthetic ClickBench `hits` table (device-native).

Shape follows the public ClickBench dataset (~100M rows of web analytics
events); only the ~25 columns the 43 queries touch are generated. Synthetic
value distributions approximate the real data's cardinalities so group-by
and LIKE selectivities are comparable. String columns are dictionary-encoded
with SORTED dictionaries (engine invariant) sized like the real data's
distinct counts (scaled); LIKE/regex evaluate over dictionary values on
device.
"""
from __future__ import annotations

import datetime as _dt
from typing import Dict

import torch

from ..engine import types as T
from ..engine.column import Column, StringColumn, Table
from .strings_gen import assemble_words, pack_vocab
from .tpch import _gen, _randint, _shard

_EPOCH = _dt.date(1970, 1, 1)
_D0 = (_dt.date(2013, 7, 1) - _EPOCH).days  # EventDate range start (spec-ish)
_DAYS = 31

_URL_WORDS = ("google", "yandex", "search", "mail", "news", "sport", "auto",
              "images", "video", "maps", "shop", "forum", "blog", "wiki",
              "music", "games", "travel", "bank", "weather", "tv")
_DOMAINS = ("example.com", "google.com", "ya.ru", "mail.ru", "news.net",
            "shop.org", "auto.io", "video.tv", "maps.app", "m.google.ru")
_PHRASE_WORDS = ("купить", "скачать", "смотреть", "онлайн", "бесплатно",
                 "google", "погода", "новости", "фото", "игры", "музыка",
                 "карта", "банк", "авто", "спорт")
_PHONE_MODELS = ["", "iPhone", "Galaxy S4", "Nokia 3310", "Xperia Z",
                 "Lumia 920", "HTC One", "Nexus 4", "Mi2", "Ascend P6"]


def _make_urls(n_distinct: int, seed: int, with_google_frac=0.08):
    """Sorted list of synthetic URLs (host-side once; becomes the device
    dictionary). A fraction contains 'google' for the LIKE queries."""
    import random

    rng = random.Random(seed)
    urls = set()
    while len(urls) < n_distinct:
        dom = rng.choice(_DOMAINS)
        parts = [rng.choice(_URL_WORDS) for _ in range(rng.randint(1, 3))]
        url = f"http://{dom}/" + "/".join(parts) + (f"?q={rng.randint(0,9999)}" if rng.random() < 0.4 else "")
        urls.add(url)
    return sorted(urls)


def _dict_string_col(n, n_distinct, seed, device, g, values=None,
                     empty_frac=0.0, skew=1.2) -> StringColumn:
    from ..engine.column import _pack_strings

    if values is None:
        values = _make_urls(n_distinct, seed)
    if empty_frac > 0 and "" not in values:
        values = [""] + list(values)
    values = sorted(set(values))
    nd = len(values)
    offs, byts = _pack_strings(values, device)
    # zipf-ish skew via squaring a uniform
    u = torch.rand(n, generator=g, device=device)
    codes = (u.pow(skew) * nd).to(torch.int32).clamp(0, nd - 1)
    if empty_frac > 0:
        empty_idx = values.index("")
        pick = torch.rand(n, generator=g, device=device) < empty_frac
        codes = torch.where(pick, torch.full_like(codes, empty_idx), codes)
    return StringColumn(offs, byts, None, codes)


def generate_hits(rows: int = 100_000_000, device="cpu", seed: int = 7,
                  rank: int = 0, world: int = 1) -> Table:
    start, n = _shard(rows, rank, world)
    dev = torch.device(device)
    g = _gen(seed, f"hits{rank}", dev)
    scale = max(rows / 100_000_000, 1e-4)

    event_date = (_D0 + _randint(0, _DAYS - 1, n, g, dev)).to(torch.int32)
    secs = _randint(0, 86399, n, g, dev)
    event_time = (event_date.to(torch.int64) * 86400 + secs) * 1_000_000

    n_urls = max(1000, int(1_000_000 * scale))
    n_phrases = max(500, int(120_000 * scale))
    n_users = max(1000, int(17_000_000 * scale))

    cols: Dict[str, Column] = {
        "WatchID": Column(T.I64, _randint(1, 1 << 60, n, g, dev)),
        "JavaEnable": Column(T.I16, _randint(0, 1, n, g, dev, torch.int16)),
        "Title": _dict_string_col(n, max(800, int(n_urls * 0.8)), seed + 1, dev, g, empty_frac=0.1),
        "GoodEvent": Column(T.I16, torch.ones(n, dtype=torch.int16, device=dev)),
        "EventTime": Column(T.TIMESTAMP, event_time),
        "EventDate": Column(T.DATE, event_date),
        "CounterID": Column(T.I32, (_randint(0, 99, n, g, dev) ** 2 % 100).to(torch.int32)),
        "ClientIP": Column(T.I32, _randint(-(1 << 31), (1 << 31) - 1, n, g, dev, torch.int32).to(torch.int32)),
        "RegionID": Column(T.I32, (_randint(0, 9999, n, g, dev) % 5000).to(torch.int32)),
        "UserID": Column(T.I64, _randint(1, n_users, n, g, dev) * 4099),
        "AdvEngineID": Column(T.I16, torch.where(_randint(0, 99, n, g, dev) < 5,
                                                 _randint(1, 60, n, g, dev),
                                                 torch.zeros(n, dtype=torch.int64, device=dev)).to(torch.int16)),
        "SearchEngineID": Column(T.I16, _randint(0, 30, n, g, dev, torch.int16).to(torch.int16)),
        "SearchPhrase": _dict_string_col(
            n, n_phrases, seed + 2, dev, g,
            values=[" ".join([_PHRASE_WORDS[i % len(_PHRASE_WORDS)],
                              _PHRASE_WORDS[(i * 7 + 3) % len(_PHRASE_WORDS)],
                              str(i)]) for i in range(n_phrases)],
            empty_frac=0.68),
        "URL": _dict_string_col(n, n_urls, seed + 3, dev, g, empty_frac=0.05),
        "Referer": _dict_string_col(n, max(500, int(n_urls * 0.3)), seed + 4, dev, g, empty_frac=0.3),
        "ResolutionWidth": Column(T.I16, (_randint(0, 20, n, g, dev) * 96 + 160).to(torch.int16)),
        "MobilePhoneModel": _dict_string_col(n, len(_PHONE_MODELS), seed + 5, dev, g,
                                             values=_PHONE_MODELS, empty_frac=0.85),
        "MobilePhone": Column(T.I16, _randint(0, 7, n, g, dev, torch.int16).to(torch.int16)),
        "IsRefresh": Column(T.I16, (_randint(0, 9, n, g, dev) == 0).to(torch.int16)),
        "DontCountHits": Column(T.I16, (_randint(0, 19, n, g, dev) == 0).to(torch.int16)),
        "IsLink": Column(T.I16, (_randint(0, 9, n, g, dev) == 0).to(torch.int16)),
        "IsDownload": Column(T.I16, (_randint(0, 49, n, g, dev) == 0).to(torch.int16)),
        "TraficSourceID": Column(T.I16, (_randint(-1, 9, n, g, dev)).to(torch.int16)),
        "RefererHash": Column(T.I64, _randint(1, 1 << 62, n, g, dev)),
        "URLHash": Column(T.I64, _randint(1, 1 << 62, n, g, dev)),
        "WindowClientWidth": Column(T.I16, (_randint(0, 20, n, g, dev) * 96).to(torch.int16)),
        "WindowClientHeight": Column(T.I16, (_randint(0, 12, n, g, dev) * 96).to(torch.int16)),
    }
    return Table(cols)


def register_clickbench(session, rows: int = 100_000_000, device=None,
                        rank: int = 0, world: int = 1, seed: int = 7):
    dev = device or session.device
    t = generate_hits(rows=rows, device=dev, seed=seed, rank=rank, world=world)
    session.catalog.register_table("hits", t, replicated=(world == 1),
                                   global_rows=rows)
    if world > 1 and getattr(session, "dist", None) is not None:
        from ..exec.distributed import sync_table_stats

        sync_table_stats(session)
    return t



def register_clickbench_parquet(session, rows: int = 100_000_000, device=None,
                                rank: int = 0, world: int = 1, seed: int = 7,
                                data_dir=None):
    """Scan-inclusive ClickBench (BASELINE config #3 "hits.parquet"):
    generate on device, persist the shard as parquet, register a scan view
    so every timed query re-reads + GPU-decodes hits from disk."""
    from .scan_swap import persist_and_swap
    from .tpch import write_tpch_parquet

    dev = device or session.device
    t = generate_hits(rows=rows, device=dev, seed=seed, rank=rank, world=world)
    session.catalog.register_table("hits", t, replicated=(world == 1),
                                   global_rows=rows)
    tables = {"hits": t}
    return persist_and_swap(
        session, tables, data_dir=data_dir,
        default_dir=f"sail_hits_{rows // 1_000_000}m", rank=rank, world=world,
        write_fn=lambda tbls, d, r: write_tpch_parquet(
            tbls, d, rank=r, page_size=1 << 20,
            row_group_rows=16_000_000))
