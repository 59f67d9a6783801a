"""Synthetic TPC-H data generator (device-native).

Generates all eight TPC-H tables directly as device tensors with the spec's
schema, cardinalities, and value distributions (TPC-H v3 clause 4.2) so query
selectivities match dbgen-shaped data. This is *synthetic* data (no dbgen
text grammar); it exists because the benchmark environment has no network to
fetch real datasets — bench.py declares `"data": "synthetic"`.

Columns never referenced by the 22 derived queries (l_comment, c_address,
ps_comment, ...) are skipped unless full=True, saving HBM for the resident
working set. Deterministic per (table, column, seed).

The per-GPU shard is rows [rank::world] of each table, matching the
engine's SPMD partitioning (exec/).
"""
from __future__ import annotations

import datetime as _dt
import os
from typing import Dict, Optional

import torch

from ..engine import types as T
from ..engine.column import Column, StringColumn, Table
from .strings_gen import assemble_words, keyed_names, pack_vocab

_EPOCH = _dt.date(1970, 1, 1)


def _d(s: str) -> int:
    y, m, dd = s.split("-")
    return (_dt.date(int(y), int(m), int(dd)) - _EPOCH).days


STARTDATE = _d("1992-01-01")
CURRENTDATE = _d("1995-06-17")
ENDDATE = _d("1998-08-02")

DEC2 = T.DecimalType(12, 2)

NATIONS = [
    ("ALGERIA", 0), ("ARGENTINA", 1), ("BRAZIL", 1), ("CANADA", 1), ("EGYPT", 4),
    ("ETHIOPIA", 0), ("FRANCE", 3), ("GERMANY", 3), ("INDIA", 2), ("INDONESIA", 2),
    ("IRAN", 4), ("IRAQ", 4), ("JAPAN", 2), ("JORDAN", 4), ("KENYA", 0),
    ("MOROCCO", 0), ("MOZAMBIQUE", 0), ("PERU", 1), ("CHINA", 2), ("ROMANIA", 3),
    ("SAUDI ARABIA", 4), ("VIETNAM", 2), ("RUSSIA", 3), ("UNITED KINGDOM", 3),
    ("UNITED STATES", 1),
]
REGIONS = ["AFRICA", "AMERICA", "ASIA", "EUROPE", "MIDDLE EAST"]

SEGMENTS = ["AUTOMOBILE", "BUILDING", "FURNITURE", "HOUSEHOLD", "MACHINERY"]
PRIORITIES = ["1-URGENT", "2-HIGH", "3-MEDIUM", "4-NOT SPECIFIED", "5-LOW"]
SHIPMODES = ["AIR", "FOB", "MAIL", "RAIL", "REG AIR", "SHIP", "TRUCK"]
INSTRUCTIONS = ["COLLECT COD", "DELIVER IN PERSON", "NONE", "TAKE BACK RETURN"]

TYPE_SYL1 = ["STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO"]
TYPE_SYL2 = ["ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"]
TYPE_SYL3 = ["TIN", "NICKEL", "BRASS", "STEEL", "COPPER"]
CONTAINER_SYL1 = ["SM", "LG", "MED", "JUMBO", "WRAP"]
CONTAINER_SYL2 = ["CASE", "BOX", "BAG", "JAR", "PKG", "PACK", "CAN", "DRUM"]

# TPC-H P_NAME color words (spec appendix, 92 entries)
COLORS = (
    "almond antique aquamarine azure beige bisque black blanched blue blush "
    "brown burlywood burnished chartreuse chiffon chocolate coral cornflower "
    "cornsilk cream cyan dark deep dim dodger drab firebrick floral forest "
    "frosted gainsboro ghost goldenrod green grey honeydew hot indian ivory "
    "khaki lace lavender lawn lemon light lime linen magenta maroon medium "
    "metallic midnight mint misty moccasin navajo navy olive orange orchid "
    "pale papaya peach peru pink plum powder puff purple red rose rosy royal "
    "saddle salmon sandy seashell sienna sky slate smoke snow spring steel "
    "thistle tomato turquoise violet wheat white yellow"
).split()

# word soup for comments; includes the tokens the derived queries grep for
COMMENT_WORDS = (
    "carefully bold final packages haggle furiously silent deposits sleep "
    "blithely regular accounts nag quickly express ideas boost slyly ironic "
    "theodolites detect above the even instructions wake according to pending "
    "foxes cajole unusual dependencies are special platelets requests among "
    "daring excuses use against dolphins sometimes busy courts across "
    "realms print permanent asymptotes customer complaints about fluffy "
    "pearls grow never close warhorses breach furious quick waters integrate "
    "along pinto beans solve enticing sauternes was ruthless multipliers "
).split()


def _stable_hash(s: str) -> int:
    h = 2166136261
    for ch in s.encode():
        h = ((h ^ ch) * 16777619) & 0xFFFFFFFF
    return h


def _gen(seed: int, tag: str, device) -> torch.Generator:
    """Per-(table,rank) generator ON the target device: SF100 tables are
    generated straight into HBM (no host staging)."""
    g = torch.Generator(device=device)
    g.manual_seed((seed * 1_000_003 + _stable_hash(tag)) & 0x7FFFFFFF)
    return g


def _randint(lo: int, hi: int, n: int, g, device, dtype=torch.int64) -> torch.Tensor:
    """Uniform in [lo, hi] inclusive, generated on the generator's device."""
    return torch.randint(lo, hi + 1, (n,), generator=g, dtype=dtype, device=device)


def _dict_col(codes: torch.Tensor, values, device) -> StringColumn:
    """Engine-wide invariant: string dictionaries are SORTED (sort and
    min/max on dict columns compare codes directly) — remap accordingly."""
    from ..engine.column import _pack_strings

    values = list(values)
    order = sorted(range(len(values)), key=lambda i: values[i])
    sorted_vals = [values[i] for i in order]
    remap = torch.empty(len(values), dtype=torch.int32, device=device)
    for new, old in enumerate(order):
        remap[old] = new
    offs, byts = _pack_strings(sorted_vals, device)
    return StringColumn(offs, byts, None, remap[codes.to(torch.int64)])


def _comment_col(n: int, g, device, min_w=5, max_w=9) -> StringColumn:
    vocab = pack_vocab(COMMENT_WORDS, device)
    k = _randint(min_w, max_w, n, g, device)
    ids = torch.randint(0, len(COMMENT_WORDS), (n, max_w), generator=g, device=device)
    return assemble_words(ids, k, vocab)


def _shard(n: int, rank: int, world: int) -> tuple:
    """Row range [start, count) of this rank's shard (block partitioning)."""
    base = n // world
    rem = n % world
    start = rank * base + min(rank, rem)
    count = base + (1 if rank < rem else 0)
    return start, count


class TpchGenerator:
    def __init__(self, sf: float = 1.0, device="cpu", seed: int = 42,
                 rank: int = 0, world: int = 1, full: bool = False):
        self.sf = sf
        self.device = torch.device(device)
        self.seed = seed
        self.rank = rank
        self.world = world
        self.full = full

    # -- cardinalities ----------------------------------------------------
    @property
    def n_supplier(self):
        return max(1, int(self.sf * 10_000))

    @property
    def n_customer(self):
        return max(1, int(self.sf * 150_000))

    @property
    def n_part(self):
        return max(1, int(self.sf * 200_000))

    @property
    def n_orders(self):
        return max(1, int(self.sf * 1_500_000))

    def generate_all(self) -> Dict[str, Table]:
        return {
            "region": self.region(),
            "nation": self.nation(),
            "supplier": self.supplier(),
            "customer": self.customer(),
            "part": self.part(),
            "partsupp": self.partsupp(),
            "orders": self.orders(),
            "lineitem": self.lineitem(),
        }

    # -- small tables (replicated on every rank) ---------------------------
    def region(self) -> Table:
        dev = self.device
        return Table({
            "r_regionkey": Column(T.I64, torch.arange(5, dtype=torch.int64, device=dev)),
            "r_name": StringColumn.from_pylist(REGIONS, device=dev, dict_encode=True),
            "r_comment": StringColumn.from_pylist(["" for _ in REGIONS], device=dev, dict_encode=False),
        })

    def nation(self) -> Table:
        dev = self.device
        return Table({
            "n_nationkey": Column(T.I64, torch.arange(25, dtype=torch.int64, device=dev)),
            "n_name": StringColumn.from_pylist([n for n, _ in NATIONS], device=dev, dict_encode=True),
            "n_regionkey": Column(T.I64, torch.tensor([r for _, r in NATIONS], dtype=torch.int64, device=dev)),
            "n_comment": StringColumn.from_pylist(["" for _ in NATIONS], device=dev, dict_encode=False),
        })

    # -- sharded tables ----------------------------------------------------
    def supplier(self) -> Table:
        n_total = self.n_supplier
        start, n = _shard(n_total, self.rank, self.world)
        dev = self.device
        g = _gen(self.seed, f"supplier{self.rank}", dev)
        keys = torch.arange(start + 1, start + n + 1, dtype=torch.int64, device=dev)
        cols = {
            "s_suppkey": Column(T.I64, keys),
            "s_name": keyed_names("Supplier#", keys),
            "s_nationkey": Column(T.I64, _randint(0, 24, n, g, dev)),
            "s_acctbal": Column(DEC2, _randint(-99999, 999999, n, g, dev)),
            "s_comment": _comment_col(n, g, dev),
        }
        cols["s_address"] = _comment_col(n, g, dev, 2, 4)
        cols["s_phone"] = self._phones(cols["s_nationkey"].data, g)
        out = {}
        for name in ["s_suppkey", "s_name", "s_address", "s_nationkey", "s_phone",
                     "s_acctbal", "s_comment"]:
            if name in cols:
                out[name] = cols[name]
        return Table(out)

    def customer(self) -> Table:
        n_total = self.n_customer
        start, n = _shard(n_total, self.rank, self.world)
        dev = self.device
        g = _gen(self.seed, f"customer{self.rank}", dev)
        keys = torch.arange(start + 1, start + n + 1, dtype=torch.int64, device=dev)
        phone_nat = _randint(0, 24, n, g, dev)
        cols = {
            "c_custkey": Column(T.I64, keys),
            "c_name": keyed_names("Customer#", keys),
            "c_nationkey": Column(T.I64, phone_nat),
            # c_phone: '<nation+10>-xxx-xxx-xxxx'; q22 reads substring(1,2)
            "c_phone": self._phones(phone_nat, g),
            "c_acctbal": Column(DEC2, _randint(-99999, 999999, n, g, dev)),
            "c_mktsegment": _dict_col(_randint(0, 4, n, g, dev), SEGMENTS, dev),
        }
        cols["c_address"] = _comment_col(n, g, dev, 2, 4)
        cols["c_comment"] = _comment_col(n, g, dev)
        order = ["c_custkey", "c_name", "c_address", "c_nationkey", "c_phone",
                 "c_acctbal", "c_mktsegment", "c_comment"]
        return Table({k: cols[k] for k in order})

    def _phones(self, nationkey: torch.Tensor, g) -> StringColumn:
        dev = self.device
        n = nationkey.shape[0]
        # country code = nationkey + 10 (2 digits), then 8 random digits
        digits = torch.empty((n, 15), dtype=torch.uint8, device=dev)
        cc = nationkey + 10
        digits[:, 0] = (torch.div(cc, 10, rounding_mode="floor") + 48).to(torch.uint8)
        digits[:, 1] = (cc % 10 + 48).to(torch.uint8)
        digits[:, 2] = 45  # '-'
        rnd = torch.randint(0, 10, (n, 12), generator=g, device=dev)
        pos = 3
        ri = 0
        for seg in (3, 3, 4):
            for _ in range(seg):
                digits[:, pos] = (rnd[:, ri] + 48).to(torch.uint8)
                pos += 1
                ri += 1
            if pos < 15:
                digits[:, pos] = 45
                pos += 1
        offsets = torch.arange(0, (n + 1) * 15, 15, dtype=torch.int64, device=dev)
        return StringColumn(offsets, digits.reshape(-1))

    def part(self) -> Table:
        n_total = self.n_part
        start, n = _shard(n_total, self.rank, self.world)
        dev = self.device
        g = _gen(self.seed, f"part{self.rank}", dev)
        keys = torch.arange(start + 1, start + n + 1, dtype=torch.int64, device=dev)
        # p_name: 5 distinct color words
        vocab = pack_vocab(COLORS, dev)
        ids = torch.randint(0, len(COLORS), (n, 5), generator=g, device=dev)
        p_name = assemble_words(ids, torch.full((n,), 5, dtype=torch.int64, device=dev), vocab)
        mfgr = _randint(1, 5, n, g, dev)
        brand = (mfgr - 1) * 5 + _randint(1, 5, n, g, dev) - 1  # 0..24
        t1 = _randint(0, 5, n, g, dev)
        t2 = _randint(0, 4, n, g, dev)
        t3 = _randint(0, 4, n, g, dev)
        types = [f"{a} {b} {c}" for a in TYPE_SYL1 for b in TYPE_SYL2 for c in TYPE_SYL3]
        type_code = (t1 * 25 + t2 * 5 + t3).to(torch.int32)
        containers = [f"{a} {b}" for a in CONTAINER_SYL1 for b in CONTAINER_SYL2]
        cont_code = _randint(0, len(containers) - 1, n, g, dev)
        retail = (90000 + (keys % 20001) // 10 * 10 + 100 * (keys % 1000) // 10)  # ~spec formula, cents
        cols = {
            "p_partkey": Column(T.I64, keys),
            "p_name": p_name,
            "p_mfgr": _dict_col(mfgr - 1, [f"Manufacturer#{i}" for i in range(1, 6)], dev),
            "p_brand": _dict_col(brand, [f"Brand#{i}{j}" for i in range(1, 6) for j in range(1, 6)], dev),
            "p_type": _dict_col(type_code, types, dev),
            "p_size": Column(T.I32, _randint(1, 50, n, g, dev, torch.int32).to(torch.int32)),
            "p_container": _dict_col(cont_code, containers, dev),
            "p_retailprice": Column(DEC2, retail),
        }
        if self.full:
            cols["p_comment"] = _comment_col(n, g, dev, 2, 5)
        return Table(cols)

    def partsupp(self) -> Table:
        # 4 suppliers per part; sharded by part
        start_p, np_ = _shard(self.n_part, self.rank, self.world)
        dev = self.device
        g = _gen(self.seed, f"partsupp{self.rank}", dev)
        n = np_ * 4
        pkeys = torch.arange(start_p + 1, start_p + np_ + 1, dtype=torch.int64, device=dev)
        ps_partkey = pkeys.repeat_interleave(4)
        i = torch.arange(n, dtype=torch.int64, device=dev) % 4
        S = self.n_supplier
        # spec supplier spread formula keeps part->supplier joins uniform
        ps_suppkey = (ps_partkey + i * (S // 4 + (ps_partkey - 1) // S)) % S + 1
        cols = {
            "ps_partkey": Column(T.I64, ps_partkey),
            "ps_suppkey": Column(T.I64, ps_suppkey),
            "ps_availqty": Column(T.I32, _randint(1, 9999, n, g, dev, torch.int32).to(torch.int32)),
            "ps_supplycost": Column(DEC2, _randint(100, 100000, n, g, dev)),
        }
        if self.full:
            cols["ps_comment"] = _comment_col(n, g, dev)
        return Table(cols)

    def orders(self) -> Table:
        start, n = _shard(self.n_orders, self.rank, self.world)
        dev = self.device
        g = _gen(self.seed, f"orders{self.rank}", dev)
        keys = torch.arange(start + 1, start + n + 1, dtype=torch.int64, device=dev)
        # customers with custkey%3==0 have no orders (spec): map onto 2/3 of keys
        C = self.n_customer
        raw = _randint(1, max(C * 2 // 3, 1), n, g, dev)
        o_custkey = torch.div(raw * 3 - 1, 2, rounding_mode="floor")  # skips multiples of 3
        odate = _randint(STARTDATE, ENDDATE - 151, n, g, dev, torch.int32).to(torch.int32)
        cols = {
            "o_orderkey": Column(T.I64, keys),
            "o_custkey": Column(T.I64, o_custkey),
            # o_orderstatus is derived from lineitems in dbgen; approximate the
            # observed mix (F ~49%, O ~49%, P ~2%)
            "o_orderstatus": _dict_col(self._status_codes(odate), ["F", "O", "P"], dev),
            "o_totalprice": Column(DEC2, _randint(100000, 50000000, n, g, dev)),
            "o_orderdate": Column(T.DATE, odate),
            "o_orderpriority": _dict_col(_randint(0, 4, n, g, dev), PRIORITIES, dev),
            "o_shippriority": Column(T.I32, torch.zeros(n, dtype=torch.int32, device=dev)),
            "o_comment": _comment_col(n, g, dev, 6, 12),
        }
        if self.full:
            cols["o_clerk"] = keyed_names("Clerk#", _randint(1, max(1, int(self.sf * 1000)), n, g, dev))
        order = ["o_orderkey", "o_custkey", "o_orderstatus", "o_totalprice", "o_orderdate",
                 "o_orderpriority", "o_clerk", "o_shippriority", "o_comment"]
        return Table({k: cols[k] for k in order if k in cols})

    def _status_codes(self, odate: torch.Tensor) -> torch.Tensor:
        cutoff = CURRENTDATE - 70
        f = odate.to(torch.int64) < cutoff
        mid = (odate.to(torch.int64) >= cutoff) & (odate.to(torch.int64) < CURRENTDATE + 30)
        return torch.where(f, torch.zeros_like(odate, dtype=torch.int64),
                           torch.where(mid, torch.full_like(odate, 2, dtype=torch.int64),
                                       torch.ones_like(odate, dtype=torch.int64)))

    def lineitem(self) -> Table:
        """~4 lineitems per order (1-7 uniform), sharded with orders so that
        l_orderkey co-partitions with o_orderkey per rank."""
        start, n_ord = _shard(self.n_orders, self.rank, self.world)
        dev = self.device
        g = _gen(self.seed, f"lineitem{self.rank}", dev)
        okeys = torch.arange(start + 1, start + n_ord + 1, dtype=torch.int64, device=dev)
        per = _randint(1, 7, n_ord, g, dev)
        n = int(per.sum().item())
        l_orderkey = okeys.repeat_interleave(per)
        # linenumber within order
        ends = torch.cumsum(per, 0)
        starts = ends - per
        l_linenumber = (torch.arange(n, dtype=torch.int64, device=dev)
                        - starts.repeat_interleave(per) + 1).to(torch.int32)
        # orderdate per order replicated to items (same distribution as orders)
        g2 = _gen(self.seed, f"orders{self.rank}", dev)
        _ = _randint(1, max(self.n_customer * 2 // 3, 1), n_ord, g2, dev)  # skip custkey draw
        odate = _randint(STARTDATE, ENDDATE - 151, n_ord, g2, dev, torch.int32).to(torch.int32)
        odate_l = odate.repeat_interleave(per).to(torch.int64)

        P = self.n_part
        S = self.n_supplier
        l_partkey = _randint(1, P, n, g, dev)
        i4 = _randint(0, 3, n, g, dev)
        l_suppkey = (l_partkey + i4 * (S // 4 + (l_partkey - 1) // S)) % S + 1
        qty = _randint(1, 50, n, g, dev)
        retail = (90000 + (l_partkey % 20001) // 10 * 10 + 100 * (l_partkey % 1000) // 10)
        extprice = qty * retail  # cents (decimal scale 2)
        disc = _randint(0, 10, n, g, dev)   # 0.00 - 0.10
        tax = _randint(0, 8, n, g, dev)     # 0.00 - 0.08
        shipdate = odate_l + _randint(1, 121, n, g, dev)
        commitdate = odate_l + _randint(30, 90, n, g, dev)
        receiptdate = shipdate + _randint(1, 30, n, g, dev)
        # returnflag: R/A if receipt <= currentdate else N
        ra = _randint(0, 1, n, g, dev)
        retcode = torch.where(receiptdate <= CURRENTDATE, ra, torch.full_like(ra, 2))
        # linestatus: O if shipdate > currentdate else F
        lscode = (shipdate > CURRENTDATE).to(torch.int64)
        cols = {
            "l_orderkey": Column(T.I64, l_orderkey),
            "l_partkey": Column(T.I64, l_partkey),
            "l_suppkey": Column(T.I64, l_suppkey),
            "l_linenumber": Column(T.I32, l_linenumber),
            "l_quantity": Column(DEC2, qty * 100),
            "l_extendedprice": Column(DEC2, extprice),
            "l_discount": Column(DEC2, disc),
            "l_tax": Column(DEC2, tax),
            "l_returnflag": _dict_col(retcode, ["A", "R", "N"], dev),
            "l_linestatus": _dict_col(lscode, ["F", "O"], dev),
            "l_shipdate": Column(T.DATE, shipdate.to(torch.int32)),
            "l_commitdate": Column(T.DATE, commitdate.to(torch.int32)),
            "l_receiptdate": Column(T.DATE, receiptdate.to(torch.int32)),
            "l_shipinstruct": _dict_col(_randint(0, 3, n, g, dev), INSTRUCTIONS, dev),
            "l_shipmode": _dict_col(_randint(0, 6, n, g, dev), SHIPMODES, dev),
        }
        if self.full:
            cols["l_comment"] = _comment_col(n, g, dev, 3, 7)
        return Table(cols)


def register_tpch(session, sf: float = 0.01, device=None, rank: int = 0, world: int = 1,
                  seed: int = 42, full: bool = False):
    """Generate TPC-H tables and register them in the session catalog."""
    dev = device or session.device
    gen = TpchGenerator(sf=sf, device=dev, seed=seed, rank=rank, world=world, full=full)
    tables = gen.generate_all()
    # planning statistics must match on every rank: use the deterministic
    # whole-table cardinalities, not the local shard size
    globals_ = {"region": 5, "nation": 25, "supplier": gen.n_supplier,
                "customer": gen.n_customer, "part": gen.n_part,
                "partsupp": gen.n_part * 4, "orders": gen.n_orders,
                "lineitem": gen.n_orders * 4}
    for name, tbl in tables.items():
        session.catalog.register_table(
            name, tbl, replicated=(world == 1 or name in ("region", "nation")),
            global_rows=globals_[name])
    if world > 1 and getattr(session, "dist", None) is not None:
        from ..exec.distributed import sync_table_stats

        sync_table_stats(session)
    return tables


# ===========================================================================
# Parquet-backed mode: write shards + register scan views so every query
# runs scan -> GPU decode -> execute (BASELINE config #2 "Parquet->Arrow in
# HBM"; ref: crates/sail-data-source/src/formats/parquet writer options).
# ===========================================================================

def write_tpch_parquet(tables: Dict[str, Table], out_dir: str, rank: int = 0,
                       row_group_rows: int = 1 << 20,
                       page_size: int = 1 << 20,
                       compression: str = "NONE") -> Dict[str, str]:
    """Write each table shard as one parquet file with GPU-decoder-friendly
    encodings: dictionary for dict-encoded strings, DELTA_LENGTH_BYTE_ARRAY
    for raw strings (fully parallel decode), DELTA_BINARY_PACKED for
    ints/dates (small-range deltas bit-pack tightly), PLAIN/FLBA otherwise.
    Returns {table: path}. Existing files with matching sizes are reused."""
    import pyarrow.parquet as pq

    from ..datasource.arrow_io import chunk_to_arrow
    from ..engine.chunk import Chunk
    from ..engine.column import StringColumn

    os.makedirs(out_dir, exist_ok=True)
    paths = {}
    for name, tbl in tables.items():
        path = os.path.join(out_dir, f"{name}-r{rank:03d}.parquet")
        paths[name] = path
        done = path + ".ok"
        if os.path.exists(path) and os.path.exists(done):
            with open(done) as f:
                if f.read().strip() == str(tbl.num_rows):
                    continue  # reuse (local dev loops; GPU boxes are fresh)
        schema = [(n, c.dtype) for n, c in tbl.columns.items()]
        use_dict = []
        col_enc = {}
        for n, c in tbl.columns.items():
            if isinstance(c, StringColumn):
                if c.is_dict:
                    use_dict.append(n)
                else:
                    col_enc[n] = "DELTA_LENGTH_BYTE_ARRAY"
            elif c.dtype.is_integer or isinstance(c.dtype, (T.DateType,
                                                            T.TimestampType,
                                                            T.DecimalType)):
                # decimals are stored as INT64 (store_decimal_as_integer)
                # so they delta-pack like any int — roughly half the FLBA
                # bytes on the wire and a plain int64 decode on the GPU
                col_enc[n] = "DELTA_BINARY_PACKED"
        n = tbl.num_rows
        writer = None
        try:
            for lo in range(0, max(n, 1), row_group_rows):
                ln = min(row_group_rows, n - lo)
                sub = Chunk([c.slice(lo, ln) for c in tbl.columns.values()],
                            [cn for cn in tbl.columns])
                at = chunk_to_arrow(sub, schema)
                if writer is None:
                    writer = pq.ParquetWriter(
                        path, at.schema, compression=compression,
                        use_dictionary=use_dict or False,
                        column_encoding=col_enc or None,
                        data_page_size=page_size,
                        store_decimal_as_integer=True,
                        # keep big dictionaries dict-encoded (ClickBench
                        # URL/Title): a PLAIN fallback mid-chunk would push
                        # the column to the host decode path
                        dictionary_pagesize_limit=64 << 20,
                        data_page_version="1.0")
                writer.write_table(at, row_group_size=row_group_rows)
                if n == 0:
                    break
        finally:
            if writer is not None:
                writer.close()
        with open(done, "w") as f:
            f.write(str(tbl.num_rows))
    return paths


def register_tpch_parquet(session, sf: float = 0.01, device=None, rank: int = 0,
                          world: int = 1, seed: int = 42, full: bool = False,
                          data_dir: Optional[str] = None):
    """Scan-inclusive TPC-H: generate on device, persist parquet shards,
    register scan VIEWS so each query re-reads + GPU-decodes its columns.
    Planner statistics are computed while the data is still resident (the
    analogue of the reference's statistics cache) and survive the swap."""
    from .scan_swap import persist_and_swap

    dev = device or session.device
    gen = TpchGenerator(sf=sf, device=dev, seed=seed, rank=rank, world=world,
                        full=full)
    tables = gen.generate_all()
    globals_ = {"region": 5, "nation": 25, "supplier": gen.n_supplier,
                "customer": gen.n_customer, "part": gen.n_part,
                "partsupp": gen.n_part * 4, "orders": gen.n_orders,
                "lineitem": gen.n_orders * 4}
    for name, tbl in tables.items():
        session.catalog.register_table(
            name, tbl, replicated=(world == 1 or name in ("region", "nation")),
            global_rows=globals_[name])
    return persist_and_swap(
        session, tables, data_dir=data_dir,
        default_dir=f"sail_tpch_sf{sf:g}", rank=rank, world=world,
        write_fn=lambda tbls, d, r: write_tpch_parquet(tbls, d, rank=r))
