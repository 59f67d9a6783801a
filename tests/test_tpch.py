"""TPC-H end-to-end: all 22 queries run, and a core subset is validated
against an independent pandas implementation on the same generated data
(the analogue of the reference's duckdb-checked TPC-H suite,
ref: python/pysail/tests/spark/test_tpch.py)."""
import datetime as dt
import math

import pandas as pd
import pytest

import sail_amd
from sail_amd.datagen.tpch import register_tpch
from sail_amd.datagen.tpch_queries import QUERIES

SF = 0.01


@pytest.fixture(scope="module")
def env():
    s = sail_amd.SessionContext(device="cpu")
    tables = register_tpch(s, sf=SF)
    dfs = {name: pd.DataFrame(t.to_pydict()) for name, t in tables.items()}
    return s, dfs


@pytest.mark.parametrize("q", list(range(1, 23)))
def test_query_runs(env, q):
    s, _ = env
    rows = s.sql(QUERIES[q]).collect()
    assert isinstance(rows, list)


def _close(a, b, tol=1e-6):
    if a is None and b is None:
        return True
    if isinstance(a, float) or isinstance(b, float):
        if isinstance(b, float) and (math.isnan(b) if isinstance(b, float) else False):
            return a is None
        return abs(float(a) - float(b)) <= tol * max(1.0, abs(float(b)))
    return a == b


def _assert_rows(got, want, tol=1e-6):
    assert len(got) == len(want), f"row count {len(got)} != {len(want)}"
    for i, (g, w) in enumerate(zip(got, want)):
        assert len(g) == len(w), f"row {i} width"
        for j, (gv, wv) in enumerate(zip(g, w)):
            assert _close(gv, wv, tol), f"row {i} col {j}: {gv!r} != {wv!r}"


def test_q1_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]
    cutoff = dt.date(1998, 12, 1) - dt.timedelta(days=90)
    d = li[li.l_shipdate <= cutoff].copy()
    d["disc_price"] = d.l_extendedprice * (1 - d.l_discount)
    d["charge"] = d.disc_price * (1 + d.l_tax)
    g = d.groupby(["l_returnflag", "l_linestatus"]).agg(
        sum_qty=("l_quantity", "sum"), sum_base_price=("l_extendedprice", "sum"),
        sum_disc_price=("disc_price", "sum"), sum_charge=("charge", "sum"),
        avg_qty=("l_quantity", "mean"), avg_price=("l_extendedprice", "mean"),
        avg_disc=("l_discount", "mean"), count_order=("l_quantity", "count"),
    ).reset_index().sort_values(["l_returnflag", "l_linestatus"])
    got = s.sql(QUERIES[1]).collect()
    want = [tuple(r) for r in g.itertuples(index=False)]
    # decimal rounding: engine keeps exact cents; pandas floats — tolerance
    _assert_rows(got, want, tol=1e-4)


def test_q3_vs_pandas(env):
    s, dfs = env
    cust = dfs["customer"]; orders = dfs["orders"]; li = dfs["lineitem"]
    c = cust[cust.c_mktsegment == "BUILDING"]
    o = orders[orders.o_orderdate < dt.date(1995, 3, 15)]
    l = li[li.l_shipdate > dt.date(1995, 3, 15)].copy()
    j = l.merge(o, left_on="l_orderkey", right_on="o_orderkey").merge(
        c, left_on="o_custkey", right_on="c_custkey")
    j["rev"] = j.l_extendedprice * (1 - j.l_discount)
    g = j.groupby(["l_orderkey", "o_orderdate", "o_shippriority"]).rev.sum().reset_index()
    g = g.sort_values(["rev", "o_orderdate"], ascending=[False, True]).head(10)
    want = [(r.l_orderkey, round(r.rev, 4), r.o_orderdate, r.o_shippriority)
            for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[3]).collect()
    _assert_rows(got, want, tol=1e-4)


def test_q4_vs_pandas(env):
    s, dfs = env
    orders = dfs["orders"]; li = dfs["lineitem"]
    o = orders[(orders.o_orderdate >= dt.date(1993, 7, 1))
               & (orders.o_orderdate < dt.date(1993, 10, 1))]
    lk = set(li[li.l_commitdate < li.l_receiptdate].l_orderkey)
    o = o[o.o_orderkey.isin(lk)]
    g = o.groupby("o_orderpriority").size().reset_index(name="n").sort_values("o_orderpriority")
    want = [(r.o_orderpriority, r.n) for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[4]).collect()
    _assert_rows(got, want)


def test_q5_vs_pandas(env):
    s, dfs = env
    j = (dfs["lineitem"]
         .merge(dfs["orders"], left_on="l_orderkey", right_on="o_orderkey")
         .merge(dfs["customer"], left_on="o_custkey", right_on="c_custkey")
         .merge(dfs["supplier"], left_on="l_suppkey", right_on="s_suppkey"))
    j = j[j.c_nationkey == j.s_nationkey]
    j = j.merge(dfs["nation"], left_on="s_nationkey", right_on="n_nationkey")
    j = j.merge(dfs["region"], left_on="n_regionkey", right_on="r_regionkey")
    j = j[(j.r_name == "ASIA") & (j.o_orderdate >= dt.date(1994, 1, 1))
          & (j.o_orderdate < dt.date(1995, 1, 1))]
    j["rev"] = j.l_extendedprice * (1 - j.l_discount)
    g = j.groupby("n_name").rev.sum().reset_index().sort_values("rev", ascending=False)
    want = [(r.n_name, round(r.rev, 4)) for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[5]).collect()
    _assert_rows(got, want, tol=1e-4)


def test_q6_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]
    d = li[(li.l_shipdate >= dt.date(1994, 1, 1)) & (li.l_shipdate < dt.date(1995, 1, 1))
           & (li.l_discount >= 0.05 - 1e-9) & (li.l_discount <= 0.07 + 1e-9)
           & (li.l_quantity < 24)]
    want = [(round(float((d.l_extendedprice * d.l_discount).sum()), 4),)]
    got = s.sql(QUERIES[6]).collect()
    _assert_rows(got, want, tol=1e-6)


def test_q12_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]; orders = dfs["orders"]
    l = li[li.l_shipmode.isin(["MAIL", "SHIP"])
           & (li.l_commitdate < li.l_receiptdate)
           & (li.l_shipdate < li.l_commitdate)
           & (li.l_receiptdate >= dt.date(1994, 1, 1))
           & (li.l_receiptdate < dt.date(1995, 1, 1))]
    j = l.merge(orders, left_on="l_orderkey", right_on="o_orderkey")
    j["high"] = j.o_orderpriority.isin(["1-URGENT", "2-HIGH"]).astype(int)
    j["low"] = 1 - j.high
    g = j.groupby("l_shipmode").agg(high=("high", "sum"), low=("low", "sum")).reset_index()
    g = g.sort_values("l_shipmode")
    want = [(r.l_shipmode, r.high, r.low) for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[12]).collect()
    _assert_rows(got, want)


def test_q13_vs_pandas(env):
    s, dfs = env
    cust = dfs["customer"]; orders = dfs["orders"]
    o = orders[~orders.o_comment.str.contains(r"special.*requests", regex=True)]
    cnt = o.groupby("o_custkey").size()
    per_cust = cust.c_custkey.map(cnt).fillna(0).astype(int)
    g = per_cust.value_counts().reset_index()
    g.columns = ["c_count", "custdist"]
    g = g.sort_values(["custdist", "c_count"], ascending=[False, False])
    want = [(r.c_count, r.custdist) for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[13]).collect()
    _assert_rows(got, want)


def test_q14_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]; part = dfs["part"]
    l = li[(li.l_shipdate >= dt.date(1995, 9, 1)) & (li.l_shipdate < dt.date(1995, 10, 1))]
    j = l.merge(part, left_on="l_partkey", right_on="p_partkey")
    j["rev"] = j.l_extendedprice * (1 - j.l_discount)
    promo = j[j.p_type.str.startswith("PROMO")].rev.sum()
    want = [(round(float(100.0 * promo / j.rev.sum()), 6),)]
    got = s.sql(QUERIES[14]).collect()
    _assert_rows(got, want, tol=1e-6)


def test_q17_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]; part = dfs["part"]
    p = part[(part.p_brand == "Brand#23") & (part.p_container == "MED BOX")]
    j = li.merge(p, left_on="l_partkey", right_on="p_partkey")
    avg_q = li.groupby("l_partkey").l_quantity.mean()
    j = j[j.l_quantity < 0.2 * j.l_partkey.map(avg_q)]
    val = j.l_extendedprice.sum() / 7.0
    got = s.sql(QUERIES[17]).collect()
    if len(j) == 0:
        assert got[0][0] is None or got == []
    else:
        _assert_rows(got, [(round(float(val), 4),)], tol=1e-4)


def test_q19_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]; part = dfs["part"]
    j = li.merge(part, left_on="l_partkey", right_on="p_partkey")
    m1 = ((j.p_brand == "Brand#12") & j.p_container.isin(["SM CASE", "SM BOX", "SM PACK", "SM PKG"])
          & (j.l_quantity >= 1) & (j.l_quantity <= 11) & j.p_size.between(1, 5))
    m2 = ((j.p_brand == "Brand#23") & j.p_container.isin(["MED BAG", "MED BOX", "MED PKG", "MED PACK"])
          & (j.l_quantity >= 10) & (j.l_quantity <= 20) & j.p_size.between(1, 10))
    m3 = ((j.p_brand == "Brand#34") & j.p_container.isin(["LG CASE", "LG BOX", "LG PACK", "LG PKG"])
          & (j.l_quantity >= 20) & (j.l_quantity <= 30) & j.p_size.between(1, 15))
    common = j.l_shipmode.isin(["AIR", "AIR REG"]) & (j.l_shipinstruct == "DELIVER IN PERSON")
    d = j[(m1 | m2 | m3) & common]
    rev = (d.l_extendedprice * (1 - d.l_discount)).sum()
    got = s.sql(QUERIES[19]).collect()
    if len(d) == 0:
        assert got[0][0] is None
    else:
        _assert_rows(got, [(round(float(rev), 4),)], tol=1e-6)


def test_q21_vs_pandas(env):
    s, dfs = env
    li = dfs["lineitem"]; orders = dfs["orders"]
    sup = dfs["supplier"]; nat = dfs["nation"]
    l1 = li[li.l_receiptdate > li.l_commitdate]
    multi = li.groupby("l_orderkey").l_suppkey.nunique()
    late_multi = l1.groupby("l_orderkey").l_suppkey.nunique()
    j = l1.merge(orders[orders.o_orderstatus == "F"], left_on="l_orderkey", right_on="o_orderkey")
    # exists: another supplier in order; not exists: another supplier late
    j = j[j.l_orderkey.map(multi).fillna(0) > 1]
    j = j[j.l_orderkey.map(late_multi).fillna(0) == 1]
    j = j.merge(sup, left_on="l_suppkey", right_on="s_suppkey")
    j = j.merge(nat, left_on="s_nationkey", right_on="n_nationkey")
    j = j[j.n_name == "SAUDI ARABIA"]
    g = j.groupby("s_name").size().reset_index(name="numwait")
    g = g.sort_values(["numwait", "s_name"], ascending=[False, True]).head(100)
    want = [(r.s_name, r.numwait) for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[21]).collect()
    _assert_rows(got, want)


def test_q22_vs_pandas(env):
    s, dfs = env
    cust = dfs["customer"]; orders = dfs["orders"]
    codes = ["13", "31", "23", "29", "30", "18", "17"]
    c = cust[cust.c_phone.str[:2].isin(codes)]
    avg_bal = c[c.c_acctbal > 0].c_acctbal.mean()
    have_orders = set(orders.o_custkey)
    d = c[(c.c_acctbal > avg_bal) & ~c.c_custkey.isin(have_orders)].copy()
    d["cc"] = d.c_phone.str[:2]
    g = d.groupby("cc").agg(n=("c_acctbal", "count"), tot=("c_acctbal", "sum")).reset_index()
    g = g.sort_values("cc")
    want = [(r.cc, r.n, round(r.tot, 4)) for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[22]).collect()
    _assert_rows(got, want, tol=1e-4)


def test_q2_vs_pandas(env):
    s, dfs = env
    p, su, ps = dfs["part"], dfs["supplier"], dfs["partsupp"]
    na, re = dfs["nation"], dfs["region"]
    eu = na.merge(re[re.r_name == "EUROPE"], left_on="n_regionkey",
                  right_on="r_regionkey")
    sup_eu = su.merge(eu, left_on="s_nationkey", right_on="n_nationkey")
    j = ps.merge(sup_eu, left_on="ps_suppkey", right_on="s_suppkey")
    min_cost = j.groupby("ps_partkey").ps_supplycost.min().rename("mc")
    pp = p[(p.p_size == 15) & p.p_type.str.endswith("BRASS")]
    full = pp.merge(j, left_on="p_partkey", right_on="ps_partkey") \
             .merge(min_cost, left_on="p_partkey", right_index=True)
    full = full[full.ps_supplycost == full.mc]
    full = full.sort_values(["s_acctbal", "n_name", "s_name", "p_partkey"],
                            ascending=[False, True, True, True]).head(100)
    want = [tuple(r) for r in full[
        ["s_acctbal", "s_name", "n_name", "p_partkey", "p_mfgr",
         "s_address", "s_phone", "s_comment"]].itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[2]).collect(), want, tol=1e-4)


def test_q7_vs_pandas(env):
    s, dfs = env
    li, o, c = dfs["lineitem"], dfs["orders"], dfs["customer"]
    su, na = dfs["supplier"], dfs["nation"]
    d = li[(li.l_shipdate >= dt.date(1995, 1, 1))
           & (li.l_shipdate <= dt.date(1996, 12, 31))]
    j = d.merge(o, left_on="l_orderkey", right_on="o_orderkey") \
         .merge(c, left_on="o_custkey", right_on="c_custkey") \
         .merge(su, left_on="l_suppkey", right_on="s_suppkey") \
         .merge(na.add_prefix("n1_"), left_on="s_nationkey",
                right_on="n1_n_nationkey") \
         .merge(na.add_prefix("n2_"), left_on="c_nationkey",
                right_on="n2_n_nationkey")
    m = (((j.n1_n_name == "FRANCE") & (j.n2_n_name == "GERMANY"))
         | ((j.n1_n_name == "GERMANY") & (j.n2_n_name == "FRANCE")))
    j = j[m].copy()
    j["l_year"] = pd.to_datetime(j.l_shipdate).dt.year
    j["volume"] = j.l_extendedprice * (1 - j.l_discount)
    g = j.groupby(["n1_n_name", "n2_n_name", "l_year"]).volume.sum() \
         .reset_index().sort_values(["n1_n_name", "n2_n_name", "l_year"])
    want = [tuple(r) for r in g.itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[7]).collect(), want, tol=1e-4)


def test_q8_vs_pandas(env):
    s, dfs = env
    li, o, c, p = dfs["lineitem"], dfs["orders"], dfs["customer"], dfs["part"]
    su, na, re = dfs["supplier"], dfs["nation"], dfs["region"]
    j = li.merge(p[p.p_type == "ECONOMY ANODIZED STEEL"],
                 left_on="l_partkey", right_on="p_partkey") \
          .merge(su, left_on="l_suppkey", right_on="s_suppkey") \
          .merge(o, left_on="l_orderkey", right_on="o_orderkey") \
          .merge(c, left_on="o_custkey", right_on="c_custkey") \
          .merge(na.add_prefix("n1_"), left_on="c_nationkey",
                 right_on="n1_n_nationkey") \
          .merge(re, left_on="n1_n_regionkey", right_on="r_regionkey") \
          .merge(na.add_prefix("n2_"), left_on="s_nationkey",
                 right_on="n2_n_nationkey")
    j = j[(j.r_name == "AMERICA")
          & (j.o_orderdate >= dt.date(1995, 1, 1))
          & (j.o_orderdate <= dt.date(1996, 12, 31))].copy()
    j["o_year"] = pd.to_datetime(j.o_orderdate).dt.year
    j["volume"] = j.l_extendedprice * (1 - j.l_discount)
    j["bra"] = j.volume.where(j.n2_n_name == "BRAZIL", 0.0)
    g = j.groupby("o_year").agg(num=("bra", "sum"), den=("volume", "sum"))
    g["mkt_share"] = g.num / g.den
    g = g.reset_index().sort_values("o_year")
    want = [(int(r.o_year), r.mkt_share) for r in g.itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[8]).collect(), want, tol=1e-4)


def test_q9_vs_pandas(env):
    s, dfs = env
    li, o, p = dfs["lineitem"], dfs["orders"], dfs["part"]
    su, ps, na = dfs["supplier"], dfs["partsupp"], dfs["nation"]
    j = li.merge(p[p.p_name.str.contains("green")],
                 left_on="l_partkey", right_on="p_partkey") \
          .merge(su, left_on="l_suppkey", right_on="s_suppkey") \
          .merge(ps, left_on=["l_suppkey", "l_partkey"],
                 right_on=["ps_suppkey", "ps_partkey"]) \
          .merge(o, left_on="l_orderkey", right_on="o_orderkey") \
          .merge(na, left_on="s_nationkey", right_on="n_nationkey")
    j = j.copy()
    j["o_year"] = pd.to_datetime(j.o_orderdate).dt.year
    j["amount"] = (j.l_extendedprice * (1 - j.l_discount)
                   - j.ps_supplycost * j.l_quantity)
    g = j.groupby(["n_name", "o_year"]).amount.sum().reset_index() \
         .sort_values(["n_name", "o_year"], ascending=[True, False])
    want = [tuple(r) for r in g.itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[9]).collect(), want, tol=1e-4)


def test_q10_vs_pandas(env):
    s, dfs = env
    li, o, c, na = dfs["lineitem"], dfs["orders"], dfs["customer"], dfs["nation"]
    oo = o[(o.o_orderdate >= dt.date(1993, 10, 1))
           & (o.o_orderdate < dt.date(1994, 1, 1))]
    j = li[li.l_returnflag == "R"] \
        .merge(oo, left_on="l_orderkey", right_on="o_orderkey") \
        .merge(c, left_on="o_custkey", right_on="c_custkey") \
        .merge(na, left_on="c_nationkey", right_on="n_nationkey").copy()
    j["rev"] = j.l_extendedprice * (1 - j.l_discount)
    g = j.groupby(["c_custkey", "c_name", "c_acctbal", "c_phone",
                   "n_name", "c_address", "c_comment"]).rev.sum() \
         .reset_index().sort_values("rev", ascending=False).head(20)
    want = [(r.c_custkey, r.c_name, r.rev, r.c_acctbal, r.n_name,
             r.c_address, r.c_phone, r.c_comment)
            for r in g.itertuples(index=False)]
    got = s.sql(QUERIES[10]).collect()
    # revenue ties make the tail order ambiguous; compare as sorted sets
    assert len(got) == len(want)
    key = lambda r: (-round(float(r[2]), 4), r[0])
    _assert_rows(sorted(got, key=key), sorted(want, key=key), tol=1e-4)


def test_q11_vs_pandas(env):
    s, dfs = env
    ps, su, na = dfs["partsupp"], dfs["supplier"], dfs["nation"]
    ger = su.merge(na[na.n_name == "GERMANY"], left_on="s_nationkey",
                   right_on="n_nationkey")
    j = ps.merge(ger, left_on="ps_suppkey", right_on="s_suppkey").copy()
    j["val"] = j.ps_supplycost * j.ps_availqty
    total = j.val.sum() * 0.0001
    g = j.groupby("ps_partkey").val.sum().reset_index()
    g = g[g.val > total].sort_values("val", ascending=False)
    want = [tuple(r) for r in g.itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[11]).collect(), want, tol=1e-4)


def test_q15_vs_pandas(env):
    s, dfs = env
    li, su = dfs["lineitem"], dfs["supplier"]
    d = li[(li.l_shipdate >= dt.date(1996, 1, 1))
           & (li.l_shipdate < dt.date(1996, 4, 1))].copy()
    d["rev"] = d.l_extendedprice * (1 - d.l_discount)
    r0 = d.groupby("l_suppkey").rev.sum().reset_index()
    mx = r0.rev.max()
    top = r0[r0.rev == mx]
    j = su.merge(top, left_on="s_suppkey", right_on="l_suppkey") \
          .sort_values("s_suppkey")
    want = [(r.s_suppkey, r.s_name, r.s_address, r.s_phone, r.rev)
            for r in j.itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[15]).collect(), want, tol=1e-4)


def test_q16_vs_pandas(env):
    s, dfs = env
    ps, p, su = dfs["partsupp"], dfs["part"], dfs["supplier"]
    bad = set(su[su.s_comment.str.contains("Customer")
                 & su.s_comment.str.contains("Complaints")
                 & su.s_comment.str.match(".*Customer.*Complaints.*")]
              .s_suppkey)
    pp = p[(p.p_brand != "Brand#45")
           & ~p.p_type.str.startswith("MEDIUM POLISHED")
           & p.p_size.isin([49, 14, 23, 45, 19, 3, 36, 9])]
    j = ps[~ps.ps_suppkey.isin(bad)].merge(
        pp, left_on="ps_partkey", right_on="p_partkey")
    g = j.groupby(["p_brand", "p_type", "p_size"]).ps_suppkey.nunique() \
         .reset_index().sort_values(
             ["ps_suppkey", "p_brand", "p_type", "p_size"],
             ascending=[False, True, True, True])
    want = [tuple(r) for r in g[["p_brand", "p_type", "p_size",
                                 "ps_suppkey"]].itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[16]).collect(), want)


def test_q18_vs_pandas(env):
    # SF0.01 has no order with sum(l_quantity) > 300, which would make the
    # oracle trivially empty — compare at threshold 150 (same plan shape)
    s, dfs = env
    li, o, c = dfs["lineitem"], dfs["orders"], dfs["customer"]
    sql = QUERIES[18].replace("> 300", "> 150")
    big = li.groupby("l_orderkey").l_quantity.sum()
    big = set(big[big > 150].index)
    j = o[o.o_orderkey.isin(big)] \
        .merge(c, left_on="o_custkey", right_on="c_custkey") \
        .merge(li, left_on="o_orderkey", right_on="l_orderkey")
    g = j.groupby(["c_name", "c_custkey", "o_orderkey", "o_orderdate",
                   "o_totalprice"]).l_quantity.sum().reset_index() \
         .sort_values(["o_totalprice", "o_orderdate"],
                      ascending=[False, True]).head(100)
    want = [tuple(r) for r in g.itertuples(index=False)]
    assert want, "oracle must be non-empty"
    _assert_rows(s.sql(sql).collect(), want, tol=1e-4)


def test_q20_vs_pandas(env):
    s, dfs = env
    li, ps, p = dfs["lineitem"], dfs["partsupp"], dfs["part"]
    su, na = dfs["supplier"], dfs["nation"]
    forest = set(p[p.p_name.str.startswith("forest")].p_partkey)
    d = li[(li.l_shipdate >= dt.date(1994, 1, 1))
           & (li.l_shipdate < dt.date(1995, 1, 1))]
    half = d.groupby(["l_partkey", "l_suppkey"]).l_quantity.sum() * 0.5
    pool = ps[ps.ps_partkey.isin(forest)].merge(
        half.rename("hq").reset_index(),
        left_on=["ps_partkey", "ps_suppkey"],
        right_on=["l_partkey", "l_suppkey"], how="left")
    pool = pool[pool.ps_availqty > pool.hq.fillna(float("inf"))]
    sups = set(pool.ps_suppkey)
    j = su[su.s_suppkey.isin(sups)].merge(
        na[na.n_name == "CANADA"], left_on="s_nationkey",
        right_on="n_nationkey").sort_values("s_name")
    want = [(r.s_name, r.s_address) for r in j.itertuples(index=False)]
    _assert_rows(s.sql(QUERIES[20]).collect(), want)
