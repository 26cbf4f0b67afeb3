"""Generator parity pins.

Golden sources:
 - the reference's committed statistics fixtures
   plugin/trino-tpch/src/main/resources/tpch/statistics/sf{0.01,1.0}/*.json
   (values inlined below — the reference tree is absent on the GPU box);
 - the public TPC-H Q1 answer @SF1 (row counts and integral sum(quantity)
   exact; monetary sums to the published 2-decimal values within
   double-vs-decimal rounding);
 - canonical dbgen first rows of lineitem/orders @SF1.
"""
import numpy as np
import pytest

import oracle

SF1_ORDERS = 1_500_000


@pytest.fixture(scope="module")
def li_sf1():
    return oracle.gen_lineitem(1.0)


def test_rowcounts():
    assert oracle.lineitem_count(1.0) == 6_001_215          # sf1.0/lineitem.json rowCount
    assert oracle.lineitem_count(0.01, 1, 15_000) == 60_175  # sf0.01/lineitem.json rowCount


def test_sf1_column_stats(li_sf1):
    li = li_sf1
    # sf1.0/lineitem.json mins/maxes/distincts
    assert (li["shipdate"].min(), li["shipdate"].max()) == (8036, 10561)
    assert len(np.unique(li["shipdate"])) == 2526
    assert (li["commitdate"].min(), li["commitdate"].max()) == (8065, 10530)
    assert len(np.unique(li["commitdate"])) == 2466
    assert (li["receiptdate"].min(), li["receiptdate"].max()) == (8038, 10591)
    assert len(np.unique(li["receiptdate"])) == 2554
    assert (li["extendedprice"].min(), li["extendedprice"].max()) == (901.0, 104949.5)
    assert len(np.unique(li["extendedprice"])) == 933_900
    assert (li["orderkey"].min(), li["orderkey"].max()) == (1, 6_000_000)
    assert len(np.unique(li["orderkey"])) == 1_500_000
    assert (li["partkey"].min(), li["partkey"].max()) == (1, 200_000)
    assert (li["quantity"].min(), li["quantity"].max()) == (1.0, 50.0)
    assert (li["discount"].min(), li["discount"].max()) == (0.0, 0.1)
    assert (li["tax"].min(), li["tax"].max()) == (0.0, 0.08)


def test_sf1_orders_stats():
    o = oracle.gen_orders(1.0)
    # sf1.0/orders.json
    assert (o["orderdate"].min(), o["orderdate"].max()) == (8035, 10440)
    assert len(np.unique(o["orderdate"])) == 2406
    assert (o["custkey"].min(), o["custkey"].max()) == (1, 149_999)
    assert len(np.unique(o["custkey"])) == 99_996
    assert (o["orderkey"].min(), o["orderkey"].max()) == (1, 6_000_000)


def test_canonical_first_rows(li_sf1):
    """Canonical dbgen SF1 output, order 1 + order 3 (public)."""
    li = {k: v[:13] for k, v in li_sf1.items()}
    # orderkey 1, line 1: partkey 155190, qty 17, extprice 21168.23,
    # disc .04, tax .02, N, O, 1996-03-13, 1996-02-12, 1996-03-22
    assert li["orderkey"][0] == 1 and li["partkey"][0] == 155190
    assert li["quantity"][0] == 17.0 and li["extendedprice"][0] == 21168.23
    assert li["discount"][0] == 0.04 and li["tax"][0] == 0.02
    assert li["returnflag"][0] == 1 and li["linestatus"][0] == 1  # N, O
    assert li["shipdate"][0] == 9568      # 1996-03-13
    assert li["commitdate"][0] == 9538    # 1996-02-12
    assert li["receiptdate"][0] == 9577   # 1996-03-22
    # order 1 partkeys for all 6 lines
    assert list(li["partkey"][:6]) == [155190, 67310, 63700, 2132, 24027, 15635]
    assert list(li["quantity"][:6]) == [17, 36, 8, 28, 24, 32]
    # order 3 (rows 7..12): first line partkey 4297 qty 45, returnflag R
    assert li["orderkey"][7] == 3 and li["partkey"][7] == 4297
    assert li["quantity"][7] == 45.0 and li["returnflag"][7] == 2  # R
    o = oracle.gen_orders(1.0, 1, 5)
    assert list(o["custkey"]) == [36901, 78002, 123314, 136777, 44485]
    assert list(o["orderkey"]) == [1, 2, 3, 4, 5]  # keys 1..7 then 32.. (sparse)
    assert o["orderdate"][0] == 9497      # 1996-01-02


Q1_EXPECTED = {
    # (rf, ls) -> (sum_qty, count, sum_base, sum_disc_price, sum_charge, avg_qty, avg_price, avg_disc)
    (0, 0): (37734107, 1478493, 56586554400.73, 53758257134.87, 55909065222.83, 25.52, 38273.13, 0.05),
    (1, 0): (991417, 38854, 1487504710.38, 1413082168.05, 1469649223.19, 25.52, 38284.47, 0.05),
    (1, 1): (74476040, 2920374, 111701729697.74, 106118230307.61, 110367043872.50, 25.50, 38249.12, 0.05),
    (2, 0): (37719753, 1478870, 56568041380.90, 53741292684.60, 55889619119.83, 25.51, 38250.85, 0.05),
}


def test_q1_sf1_answer(li_sf1):
    """TPC-H official Q1 answer @SF1. Integral values exact; monetary sums to
    the published cents within double accumulation wobble (<1.0 absolute on
    ~1e11, i.e. <1e-11 relative)."""
    for r, name in ((oracle.q1_naive(li_sf1), "naive"),
                    (oracle.q1_exact(li_sf1, threads=4)[0], "exact")):
        seen = 0
        for c in range(6):
            if r.count[c] == 0:
                continue
            combo = (c // 2, c % 2)
            exp = Q1_EXPECTED[combo]
            assert r.sum_qty[c] == exp[0], (name, combo)
            assert r.count[c] == exp[1], (name, combo)
            assert abs(r.sum_base[c] - exp[2]) < 1.0
            assert abs(r.sum_disc_price[c] - exp[3]) < 1.0
            assert abs(r.sum_charge[c] - exp[4]) < 1.0
            assert abs(r.avg_qty[c] - exp[5]) < 0.005
            assert abs(r.avg_price[c] - exp[6]) < 0.005
            assert abs(r.avg_disc[c] - exp[7]) < 0.005
            seen += 1
        assert seen == 4


def test_part_seek_consistency():
    """Split generation (TpchSplit part/partCount semantics) must be
    bit-identical to contiguous generation — pins the log-time seed skip."""
    full = oracle.gen_lineitem(0.01)
    parts = []
    bounds = [1, 4000, 9001, 15001]  # order ranges
    for s, e in zip(bounds[:-1], bounds[1:]):
        parts.append(oracle.gen_lineitem(0.01, s, e - s))
    for col in full:
        got = np.concatenate([p[col] for p in parts])
        assert np.array_equal(full[col], got), col


def test_naive_vs_exact_wobble(li_sf1):
    """Documents DESIGN.md §6: the reference's sequential double sums differ
    from the correctly-rounded exact sums by many ULPs at scale — the reason
    the full-size parity bar is exact-sum equality, not 1 ULP vs naive."""
    rn = oracle.q1_naive(li_sf1)
    re, _ = oracle.q1_exact(li_sf1)
    max_ulps = 0
    for c in range(6):
        if rn.count[c] == 0:
            continue
        a, b = rn.sum_base[c], re.sum_base[c]
        ulp = np.spacing(b)
        max_ulps = max(max_ulps, abs(a - b) / ulp)
        assert abs(a - b) / abs(b) < 1e-9   # still tiny in relative terms
    assert max_ulps >= 1.0  # naive is NOT within 1 ULP of exact at SF1


class TestShipmodeStream:
    """L_SMODE seed 675466456 pinned by the reference's own fixtures
    (tests/golden/ref_fixtures.json, extracted by tools/extract_ref_goldens.py
    from the 785 canonical SF1 lineitem rows + the Q12 answer)."""

    MODES = ["REG AIR", "AIR", "RAIL", "TRUCK", "MAIL", "FOB", "SHIP"]

    @pytest.fixture(scope="class")
    def fixtures(self):
        import json, os
        p = os.path.join(os.path.dirname(__file__), "golden", "ref_fixtures.json")
        return json.load(open(p))

    def test_canonical_rows_all_columns(self, fixtures):
        rows = fixtures["lineitem_canonical_sf1"]["rows"]
        cols = oracle.gen_lineitem(1.0, 1, 300)
        idx = {(int(o), int(l)): i
               for i, (o, l) in enumerate(zip(cols["orderkey"], cols["linenumber"]))}
        import datetime
        epoch = datetime.date(1970, 1, 1)
        for r in rows:
            i = idx[(r["orderkey"], r["linenumber"])]
            assert self.MODES[cols["shipmode"][i]] == r["shipmode"]
            assert int(cols["quantity"][i]) == r["quantity"]
            assert abs(cols["extendedprice"][i] - float(r["extendedprice"])) < 5e-3
            assert abs(cols["discount"][i] - float(r["discount"])) < 1e-9
            assert (epoch + datetime.timedelta(days=int(cols["shipdate"][i])
                                               )).isoformat() == r["shipdate"]
            assert (epoch + datetime.timedelta(days=int(cols["receiptdate"][i])
                                               )).isoformat() == r["receiptdate"]
            assert "ANR"[cols["returnflag"][i]] == r["returnflag"]

    def test_q12_answer_exact(self, fixtures):
        exp = {r["shipmode"]: (r["high"], r["low"]) for r in fixtures["q12_sf1"]["rows"]}
        cols = oracle.gen_lineitem(1.0, columns=["orderkey", "shipmode", "shipdate",
                                                 "commitdate", "receiptdate"])
        orders = oracle.gen_orders(1.0)
        pri = dict(zip(orders["orderkey"].tolist(), orders["orderpriority"].tolist()))
        d94, d95 = 8766, 9131
        m = ((np.isin(cols["shipmode"], [4, 6])) &
             (cols["commitdate"] < cols["receiptdate"]) &
             (cols["shipdate"] < cols["commitdate"]) &
             (cols["receiptdate"] >= d94) & (cols["receiptdate"] < d95))
        got = {}
        for mode_id, name in ((4, "MAIL"), (6, "SHIP")):
            sel = m & (cols["shipmode"] == mode_id)
            oks = cols["orderkey"][sel]
            highs = sum(1 for ok in oks.tolist() if pri[ok] <= 1)
            got[name] = (highs, int(sel.sum()) - highs)
        assert got == exp


class TestCustomerStreams:
    """C_NKEY (1489529863) and C_ABAL (298370230) pinned by the Q10 SF1
    answer fixture's 20 (custkey -> nation, acctbal) rows; the phone country
    code (nationkey+10) cross-checks every row. Q22's full answer verifies
    both streams end-to-end (tests/test_gpu_q22.py on device; here the
    oracle composition)."""

    NATIONS = ["ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
               "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ",
               "JAPAN", "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU",
               "CHINA", "ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA",
               "UNITED KINGDOM", "UNITED STATES"]

    def test_q22_answer_exact(self):
        import json, os
        fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                         "ref_fixtures.json")))
        c = oracle.gen_customer(1.0)
        orders = oracle.gen_orders(1.0)
        has_order = np.zeros(150_001, bool)
        has_order[np.unique(orders["custkey"])] = True
        cc = c["nationkey"].astype(int) + 10
        codes = [13, 17, 18, 23, 29, 30, 31]
        sel = np.isin(cc, codes)
        pos = sel & (c["acctbal_cents"] > 0)
        total, npos = int(c["acctbal_cents"][pos].sum()), int(pos.sum())
        qual = sel & (c["acctbal_cents"] * npos > total) & ~has_order[c["custkey"]]
        exp = fx["all_answers_sf1"]["rows"]["q22"]
        for i, code in enumerate(codes):
            m = qual & (cc == code)
            assert int(m.sum()) == int(exp[i][1])
            assert int(c["acctbal_cents"][m].sum()) == round(float(exp[i][2]) * 100)


def test_suppkey_bridge_canonical_rows():
    """l_suppkey (L_SKEY 2095021727 + partsupp bridge) vs the 785 canonical
    rows in tests/golden/ref_fixtures.json."""
    import json, os
    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))
    cols = oracle.gen_lineitem(1.0, 1, 300)
    # the canonical csv's suppkey column (order 1, lines 1-6) — seed + bridge
    # were verified against all 785 rows at pin time; this guards the stream
    assert cols["suppkey"][:6].tolist() == [7706, 7311, 3701, 4633, 1534, 638]
    assert fx["lineitem_canonical_sf1"]["rows"][0]["partkey"] == 155190


class TestNationJoinQueries:
    """Oracle compositions of Q5/Q7/Q8 against the reference's SF1 answer
    fixtures — pins the supplier nationkey stream and the fixed nation/region
    tables end-to-end on CPU (device pipelines: tests/test_gpu_q{5,7,8}.py)."""

    @pytest.fixture(scope="class")
    def data(self):
        c = oracle.gen_customer(1.0)
        o = oracle.gen_orders(1.0)
        li = oracle.gen_lineitem(1.0, columns=["orderkey", "suppkey", "partkey",
                                               "extendedprice", "discount",
                                               "shipdate"])
        s = oracle.gen_supplier(1.0)
        cnk = np.zeros(150_001, np.int8)
        cnk[c["custkey"]] = c["nationkey"]
        snk = np.zeros(10_001, np.int8)
        snk[s["suppkey"]] = s["nationkey"]
        max_ok = int(o["orderkey"].max())
        o_cn = np.full(max_ok + 1, -1, np.int8)
        o_cn[o["orderkey"]] = cnk[o["custkey"]]
        o_dt = np.zeros(max_ok + 1, np.int32)
        o_dt[o["orderkey"]] = o["orderdate"]
        rev = li["extendedprice"] * (1.0 - li["discount"])
        return dict(li=li, snk=snk, o_cn=o_cn, o_dt=o_dt, rev=rev)

    @pytest.fixture(scope="class")
    def answers(self):
        import json, os
        return json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                           "ref_fixtures.json")))["all_answers_sf1"]["rows"]

    def test_q5(self, data, answers):
        ASIA = np.array([8, 9, 12, 18, 21])
        li = data["li"]
        cn = data["o_cn"][li["orderkey"]]
        od = data["o_dt"][li["orderkey"]]
        sn = data["snk"][li["suppkey"]]
        m = (np.isin(cn, ASIA) & (cn == sn) & (od >= 8766) & (od < 9131))
        got = {}
        for nk in ASIA:
            got[int(nk)] = data["rev"][m & (cn == nk)].sum()
        names = ["ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
                 "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ",
                 "JAPAN", "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU",
                 "CHINA", "ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA",
                 "UNITED KINGDOM", "UNITED STATES"]
        exp = {row[0]: float(row[1]) for row in answers["q05"]}
        for nk, v in got.items():
            assert abs(v - exp[names[nk]]) < 5e-4

    def test_q7(self, data, answers):
        li = data["li"]
        cn = data["o_cn"][li["orderkey"]]
        sn = data["snk"][li["suppkey"]]
        sd = li["shipdate"]
        win = (sd >= 9131) & (sd < 9862)
        year = np.where(sd < 9496, 1995, 1996)
        for i, row in enumerate(answers["q07"]):
            s_nk = 6 if row[0] == "FRANCE" else 7
            c_nk = 13 - s_nk
            m = win & (sn == s_nk) & (cn == c_nk) & (year == int(row[2]))
            assert abs(data["rev"][m].sum() - float(row[3])) < 5e-4

    def test_q8(self, data, answers):
        p = oracle.gen_part(1.0)
        ptype = np.zeros(200_001, np.int16)
        ptype[p["partkey"]] = p["type_id"]
        li = data["li"]
        cn = data["o_cn"][li["orderkey"]]
        od = data["o_dt"][li["orderkey"]]
        sn = data["snk"][li["suppkey"]]
        AMERICA = np.array([1, 2, 3, 17, 24])
        m = ((ptype[li["partkey"]] == 103) & np.isin(cn, AMERICA) &
             (od >= 9131) & (od < 9862))
        year = np.where(od < 9496, 1995, 1996)
        for i, row in enumerate(answers["q08"]):
            ym = m & (year == int(row[0]))
            share = data["rev"][ym & (sn == 2)].sum() / data["rev"][ym].sum()
            assert abs(share - float(row[1])) < 5e-7
