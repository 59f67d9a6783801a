"""Property-based differential testing: random data + random queries, engine
vs an independent pandas oracle (the reference relies on Spark's test corpus
for conformance; this is the in-repo analogue)."""
import math

import pandas as pd
import pytest
from hypothesis import given, settings, strategies as st

import sail_amd


def _mk(data):
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe(data, name="t")
    return s


rows_st = st.integers(min_value=1, max_value=60)


@st.composite
def table_st(draw):
    n = draw(rows_st)
    ints = draw(st.lists(st.one_of(st.none(), st.integers(-1000, 1000)),
                         min_size=n, max_size=n))
    floats = draw(st.lists(st.one_of(st.none(),
                                     st.floats(-1e6, 1e6, allow_nan=False)),
                           min_size=n, max_size=n))
    cats = draw(st.lists(st.sampled_from(["a", "b", "c", "dd"]),
                         min_size=n, max_size=n))
    return {"i": ints, "f": floats, "c": cats}


@settings(max_examples=30, deadline=None)
@given(table_st(), st.integers(-1000, 1000))
def test_filter_count_sum_matches_pandas(data, threshold):
    s = _mk(data)
    got = s.sql(f"SELECT count(*), count(i), sum(i) FROM t WHERE i > {threshold}").collect()[0]
    df = pd.DataFrame(data)
    sel = df[df["i"].notna() & (df["i"] > threshold)]
    want_cnt = len(sel)
    want_sum = int(sel["i"].sum()) if want_cnt else None
    assert got[0] == want_cnt and got[1] == want_cnt
    assert got[2] == want_sum


@settings(max_examples=30, deadline=None)
@given(table_st())
def test_group_agg_matches_pandas(data):
    s = _mk(data)
    got = dict((r[0], (r[1], r[2], r[3])) for r in s.sql(
        "SELECT c, count(*), sum(i), max(f) FROM t GROUP BY c").collect())
    df = pd.DataFrame(data)
    for key, grp in df.groupby("c"):
        cnt = len(grp)
        sm = int(grp["i"].sum()) if grp["i"].notna().any() else None
        mx = grp["f"].max() if grp["f"].notna().any() else None
        g = got[key]
        assert g[0] == cnt and g[1] == sm
        if mx is None:
            assert g[2] is None
        else:
            assert g[2] == pytest.approx(mx, rel=1e-12)


@settings(max_examples=20, deadline=None)
@given(table_st(), table_st())
def test_join_matches_pandas(left, right):
    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe(left, name="l")
    s.create_dataframe(right, name="r")
    got = s.sql("SELECT l.i FROM l JOIN r ON l.i = r.i").collect()
    # SQL equi-join never matches NULL keys; make the pandas oracle agree
    dl = pd.DataFrame({"i": [v for v in left["i"] if v is not None]})
    dr = pd.DataFrame({"i": [v for v in right["i"] if v is not None]})
    want = dl.merge(dr, on="i")["i"] if len(dl) and len(dr) else []
    assert sorted(x[0] for x in got) == sorted(int(v) for v in want)


@settings(max_examples=20, deadline=None)
@given(table_st(), st.sampled_from(["i", "f", "c"]),
       st.booleans(), st.integers(1, 10))
def test_order_limit_matches_pandas(data, col, asc, k):
    s = _mk(data)
    direction = "ASC" if asc else "DESC"
    got = [r[0] for r in s.sql(
        f"SELECT {col} FROM t ORDER BY {col} {direction} NULLS LAST, i, f, c "
        f"LIMIT {k}").collect()]
    df = pd.DataFrame(data)
    want = df.sort_values([col, "i", "f", "c"],
                          ascending=[asc, True, True, True],
                          na_position="last")[col].head(len(got)).tolist()
    for g, w in zip(got, want):
        if w is None or (isinstance(w, float) and math.isnan(w)):
            assert g is None
        elif isinstance(w, float):
            assert g == pytest.approx(w, rel=1e-12)
        else:
            assert g == w


@given(st.lists(st.integers(min_value=0, max_value=2**40), max_size=200))
@settings(max_examples=30, deadline=None)
def test_roaring_bitmap_round_trip(vals):
    from sail_amd.utils.roaring import roaring64_deserialize, roaring64_serialize

    out = roaring64_deserialize(roaring64_serialize(vals))
    assert out.tolist() == sorted(set(vals))


@given(st.text(max_size=40))
@settings(max_examples=30, deadline=None)
def test_url_encode_decode_round_trip(text):
    from urllib.parse import quote_plus

    s = sail_amd.SessionContext(device="cpu")
    s.create_dataframe({"t": [text if text else "x"]}, name="urlrt")
    (enc,), = s.sql("SELECT url_encode(t) FROM urlrt").collect()
    (dec,), = s.sql("SELECT url_decode(url_encode(t)) FROM urlrt").collect()
    assert enc == quote_plus(text if text else "x")
    assert dec == (text if text else "x")


@given(st.integers(min_value=-10**15, max_value=10**15),
       st.integers(min_value=-5000, max_value=5000))
@settings(max_examples=30, deadline=None)
def test_timestampadd_hours_matches_datetime(us, hours):
    import datetime as dt

    s = sail_amd.SessionContext(device="cpu")
    base = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=us)
    lit = base.strftime("%Y-%m-%d %H:%M:%S.%f")
    (got,), = s.sql(f"SELECT timestampadd(HOUR, {hours}, "
                    f"TIMESTAMP '{lit}')").collect()
    delta = (base + dt.timedelta(hours=hours)) - dt.datetime(1970, 1, 1)
    # integer-exact oracle (float total_seconds() loses sub-µs precision)
    want = (delta.days * 86_400 + delta.seconds) * 1_000_000 \
        + delta.microseconds
    assert got == want


# -- exact string keys (kill the FNV-collision class) -----------

def _mk_session():
    import sail_amd

    return sail_amd.SessionContext(device="cpu")


def test_exact_string_codes_randomized():
    import random

    from sail_amd.engine.column import StringColumn
    from sail_amd.engine.joins import exact_string_codes

    rng = random.Random(11)
    vals = [f"string-{rng.randint(0, 300)}-{'pad' * rng.randint(0, 6)}"
            for _ in range(5000)]
    col = StringColumn.from_pylist(vals, device="cpu", dict_encode=False)
    codes = exact_string_codes([col])[0].tolist()
    by_code = {}
    for v, c in zip(vals, codes):
        by_code.setdefault(c, set()).add(v)
    # each code maps to exactly one string and vice versa
    assert all(len(s) == 1 for s in by_code.values())
    assert len(by_code) == len(set(vals))


def test_group_by_exact_under_forced_h1_collision(monkeypatch):
    """All h1 values collide; grouping must still be exact (h2 + byte
    verification carry it)."""
    import torch

    from sail_amd.engine import joins

    real_pair = joins.string_hash_pair

    def collide_h1(c):
        h1, h2 = real_pair(c)
        return torch.zeros_like(h1), h2

    monkeypatch.setattr(joins, "string_hash_pair", collide_h1)
    s = _mk_session()
    vals = [f"longish-key-value-{i % 37}" for i in range(1000)]
    s.create_dataframe({"k": vals, "v": [1] * 1000}, name="fc1")
    rows = s.sql("SELECT k, count(*) FROM fc1 GROUP BY k").collect()
    assert len(rows) == 37
    assert all(r[1] == (28 if int(r[0].split("-")[-1]) < 1000 % 37 else 27)
               for r in rows)


def test_group_by_exact_under_double_collision(monkeypatch):
    """Both hash families collide (forced): detection kicks in and the host
    fallback still produces exact groups."""
    import torch

    from sail_amd.engine import joins

    def collide_both(c):
        z = torch.zeros(len(c), dtype=torch.int64)
        return z, z.clone()

    monkeypatch.setattr(joins, "string_hash_pair", collide_both)
    s = _mk_session()
    vals = [f"another-long-key-{i % 11}" for i in range(330)]
    s.create_dataframe({"k": vals, "v": [2] * 330}, name="fc2")
    rows = s.sql("SELECT k, count(*) FROM fc2 GROUP BY k").collect()
    assert len(rows) == 11 and all(r[1] == 30 for r in rows)


def test_join_exact_under_forced_h1_collision(monkeypatch):
    import torch

    from sail_amd.engine import joins

    real_pair = joins.string_hash_pair

    def collide_h1(c):
        h1, h2 = real_pair(c)
        return torch.full_like(h1, 7), h2

    monkeypatch.setattr(joins, "string_hash_pair", collide_h1)
    s = _mk_session()
    left = [f"join-key-string-{i}" for i in range(200)]
    right = [f"join-key-string-{i}" for i in range(100, 300)]
    s.create_dataframe({"k": left, "a": list(range(200))}, name="jl")
    s.create_dataframe({"k": right, "b": list(range(200))}, name="jr")
    rows = s.sql("SELECT jl.k, a, b FROM jl JOIN jr ON jl.k = jr.k").collect()
    assert len(rows) == 100  # exactly the overlap, no hash-merged extras
    for k, a, b in rows:
        assert k == f"join-key-string-{a}" and a == b + 100
