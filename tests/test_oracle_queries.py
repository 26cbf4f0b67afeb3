"""CPU oracle compositions of the round-2 TPC-H queries vs the reference's
own SF1 answer fixtures (testing/trino-product-tests/.../hive_tpch/
qNN.result, in tests/golden/ref_fixtures.json). These pin the SEMANTICS of
each query plan (incl. the BBB supplier-comment overlay via Q16, the text
pool via Q13, NULL-correlated-sum exclusion via Q20, and the min/max
supplier derivation for Q21's EXISTS/NOT EXISTS) before the device
pipelines implement the same plans through the operator C-ABI.

Q2/Q9/Q10/Q11/Q13/Q16/Q17/Q19/Q20/Q21 (round-2 queries) plus
Q3/Q4/Q6/Q14/Q15/Q18 (round-1 queries, CPU-composed at the bottom so the
whole 22-query semantic set is verifiable without a GPU — the remaining
q1/q5/q7/q8/q12/q22 live in test_tpchgen_oracle.py).
"""
import collections
import ctypes
import datetime
import json
import os

import numpy as np
import pytest

import oracle

HERE = os.path.dirname(__file__)


def D(y, m, d):
    return (datetime.date(y, m, d) - datetime.date(1970, 1, 1)).days


@pytest.fixture(scope="module")
def fx():
    with open(os.path.join(HERE, "golden", "ref_fixtures.json")) as f:
        return json.load(f)["all_answers_sf1"]["rows"]


@pytest.fixture(scope="module")
def part():
    return oracle.gen_part2(1.0)


@pytest.fixture(scope="module")
def partsupp():
    return oracle.gen_partsupp(1.0)


@pytest.fixture(scope="module")
def supplier():
    return oracle.gen_supplier2(1.0)


@pytest.fixture(scope="module")
def orders():
    return oracle.gen_orders3(1.0)


@pytest.fixture(scope="module")
def strings_lib():
    lib = ctypes.CDLL(os.path.join(HERE, "..", "trino_amd", "libtrino_gpu.so"))
    for fn in (lib.tg_tpch_supplier_strings, lib.tg_tpch_customer_strings):
        fn.restype = ctypes.c_int
        fn.argtypes = [ctypes.c_double, ctypes.c_void_p, ctypes.c_int32,
                       ctypes.c_int32] + [ctypes.c_void_p] * 6
    return lib


def supplier_strings(lib, keys, want=("name", "address", "phone", "comment")):
    keys = np.asarray(keys, np.int64)
    n = len(keys)
    bufs = {w: ctypes.create_string_buffer(n * 160) for w in want}
    lib.tg_tpch_supplier_strings(
        1.0, keys.ctypes.data, n, 160,
        bufs.get("name"), bufs.get("address"), bufs.get("phone"),
        bufs.get("comment"), None, None)
    out = {}
    for w in want:
        out[w] = [bufs[w].raw[i * 160:(i + 1) * 160].split(b"\0")[0].decode()
                  for i in range(n)]
    return out


def test_q02(fx, part, partsupp, supplier, strings_lib):
    p, ps, su = part, partsupp, supplier
    NR = np.array(oracle.NATION_REGION)
    keep = (p["size"] == 15) & ((p["type_id"] % 5) == 2)       # %BRASS
    pk_keep = set(p["partkey"][keep].tolist())
    eu = set(su["suppkey"][NR[su["nationkey"]] == 3].tolist())  # EUROPE
    best = collections.defaultdict(lambda: 1 << 60)
    for k, s, c in zip(ps["partkey"].tolist(), ps["suppkey"].tolist(),
                       ps["supplycost_cents"].tolist()):
        if k in pk_keep and s in eu and c < best[k]:
            best[k] = c
    nk = dict(zip(su["suppkey"].tolist(), su["nationkey"].tolist()))
    ab = dict(zip(su["suppkey"].tolist(), su["acctbal_cents"].tolist()))
    rows = []
    for k, s, c in zip(ps["partkey"].tolist(), ps["suppkey"].tolist(),
                       ps["supplycost_cents"].tolist()):
        if k in best and s in eu and c == best[k]:
            rows.append((ab[s], oracle.NATIONS[nk[s]], s, k))
    rows.sort(key=lambda r: (-r[0], r[1], r[2], r[3]))
    rows = rows[:100]
    exp = fx["q02"]
    assert len(rows) == len(exp)
    st = supplier_strings(strings_lib, [r[2] for r in rows])
    mfgr = dict(zip(p["partkey"].tolist(), (p["brand"] // 10).tolist()))
    for i, (r, e) in enumerate(zip(rows, exp)):
        assert abs(r[0] / 100.0 - float(e[0])) < 1e-9
        assert f"Supplier#{r[2]:09d}" == e[1] and r[1] == e[2]
        assert r[3] == int(e[3]) and f"Manufacturer#{mfgr[r[3]]}" == e[4]
        assert st["address"][i] == e[5] and st["phone"][i] == e[6]
        assert st["comment"][i] == e[7]


def test_q09(fx, part, partsupp, supplier, orders):
    p, ps, su, o = part, partsupp, supplier, orders
    green = (p["name_ids"] == oracle.COLOR_GREEN).any(axis=1)
    pkg = np.zeros(200001, bool)
    pkg[p["partkey"][green]] = True
    li = oracle.gen_lineitem(1.0, columns=["orderkey", "partkey", "suppkey",
                                           "quantity", "extendedprice",
                                           "discount"])
    m = pkg[li["partkey"]]
    cost = {(k, s): c / 100.0 for k, s, c in
            zip(ps["partkey"].tolist(), ps["suppkey"].tolist(),
                ps["supplycost_cents"].tolist())}
    nk = np.zeros(10001, np.int32)
    nk[su["suppkey"]] = su["nationkey"]
    jan = np.array([D(y, 1, 1) for y in range(1992, 2000)])
    ybo = dict(zip(o["orderkey"].tolist(),
                   (1991 + np.searchsorted(jan, o["orderdate"], side="right")).tolist()))
    agg = collections.defaultdict(float)
    for k, s, okx, q, ep, dc in zip(li["partkey"][m].tolist(), li["suppkey"][m].tolist(),
                                    li["orderkey"][m].tolist(), li["quantity"][m].tolist(),
                                    li["extendedprice"][m].tolist(), li["discount"][m].tolist()):
        agg[(int(nk[s]), ybo[okx])] += ep * (1.0 - dc) - cost[(k, s)] * q
    rows = sorted(((oracle.NATIONS[n], y, v) for (n, y), v in agg.items()),
                  key=lambda r: (r[0], -r[1]))
    exp = [(r[0], int(r[1]), float(r[2])) for r in fx["q09"]]
    assert len(rows) == len(exp)
    for a, b in zip(rows, exp):
        assert a[0] == b[0] and a[1] == b[1] and abs(a[2] - b[2]) < 0.002


def test_q10(fx, orders, strings_lib):
    o = orders
    li = oracle.gen_lineitem(1.0, columns=["orderkey", "extendedprice",
                                           "discount", "returnflag"])
    ow = (o["orderdate"] >= D(1993, 10, 1)) & (o["orderdate"] < D(1994, 1, 1))
    ck = dict(zip(o["orderkey"][ow].tolist(), o["custkey"][ow].tolist()))
    m = li["returnflag"] == 2
    rev = collections.defaultdict(float)
    for okx, ep, dc in zip(li["orderkey"][m].tolist(), li["extendedprice"][m].tolist(),
                           li["discount"][m].tolist()):
        c = ck.get(okx)
        if c is not None:
            rev[c] += ep * (1.0 - dc)
    rows = sorted(rev.items(), key=lambda kv: (-kv[1], kv[0]))[:20]
    exp = fx["q10"]
    keys = np.array([k for k, _ in rows], np.int64)
    n = len(keys)
    name = ctypes.create_string_buffer(n * 160)
    addr = ctypes.create_string_buffer(n * 160)
    ph = ctypes.create_string_buffer(n * 160)
    cm = ctypes.create_string_buffer(n * 160)
    ab = np.zeros(n, np.int64)
    nk = np.zeros(n, np.int32)
    strings_lib.tg_tpch_customer_strings(1.0, keys.ctypes.data, n, 160, name,
                                         addr, ph, cm, ab.ctypes.data,
                                         nk.ctypes.data)
    for i, ((k, v), e) in enumerate(zip(rows, exp)):
        assert k == int(e[0])
        assert name.raw[i * 160:(i + 1) * 160].split(b"\0")[0].decode() == e[1]
        assert abs(v - float(e[2])) < 0.002
        assert abs(ab[i] / 100.0 - float(e[3])) < 1e-9
        assert oracle.NATIONS[nk[i]] == e[4]
        assert addr.raw[i * 160:(i + 1) * 160].split(b"\0")[0].decode() == e[5]
        assert ph.raw[i * 160:(i + 1) * 160].split(b"\0")[0].decode() == e[6]
        assert cm.raw[i * 160:(i + 1) * 160].split(b"\0")[0].decode() == e[7]


def test_q11(fx, partsupp, supplier):
    ps, su = partsupp, supplier
    de = np.isin(ps["suppkey"], su["suppkey"][su["nationkey"] == 7])
    val = ps["supplycost_cents"][de] / 100.0 * ps["availqty"][de]
    agg = collections.defaultdict(float)
    for k, v in zip(ps["partkey"][de].tolist(), val.tolist()):
        agg[k] += v
    thr = sum(agg.values()) * 0.0001
    rows = sorted(((k, v) for k, v in agg.items() if v > thr),
                  key=lambda r: (-r[1], r[0]))
    exp = [(int(r[0]), float(r[1])) for r in fx["q11"]]
    assert len(rows) == len(exp)
    for a, b in zip(rows, exp):
        assert a[0] == b[0] and abs(a[1] - b[1]) < 0.005


def test_q13(fx, orders):
    o = orders
    pool = oracle.text_pool_bytes().tobytes()
    off, ln = o["cmnt_off"], o["cmnt_len"]
    match = np.zeros(len(off), bool)
    for i in range(len(off)):
        s = pool[off[i]:off[i] + ln[i]]
        j = s.find(b"special")
        match[i] = j >= 0 and s.find(b"requests", j + 7) >= 0
    cnt = collections.Counter(o["custkey"][~match].tolist())
    hist = collections.Counter(cnt.values())
    hist[0] = 150000 - len(cnt)
    got = sorted(hist.items(), key=lambda kv: (-kv[1], -kv[0]))
    assert got == [(int(r[0]), int(r[1])) for r in fx["q13"]]


def test_q16(fx, part, partsupp):
    p, ps = part, partsupp
    cm = oracle.gen_supplier_comments(1, 10000)
    bad = set(i + 1 for i, c in enumerate(cm)
              if "Customer" in c and "Complaints" in c[c.find("Customer"):])
    T1 = ["STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO"]
    T2 = ["ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"]
    T3 = ["TIN", "NICKEL", "BRASS", "STEEL", "COPPER"]
    keep = ((p["brand"] != 45) & np.isin(p["size"], [49, 14, 23, 45, 19, 3, 36, 9])
            & ~((p["type_id"] >= 65) & (p["type_id"] <= 69)))   # MEDIUM POLISHED%
    attrs = {int(p["partkey"][i]): (int(p["brand"][i]), int(p["type_id"][i]),
                                    int(p["size"][i]))
             for i in np.nonzero(keep)[0]}
    groups = collections.defaultdict(set)
    for k, s in zip(ps["partkey"].tolist(), ps["suppkey"].tolist()):
        a = attrs.get(k)
        if a is not None and s not in bad:
            groups[a].add(s)
    rows = [(f"Brand#{br}", f"{T1[ty//25]} {T2[(ty//5)%5]} {T3[ty%5]}", sz, len(s))
            for (br, ty, sz), s in groups.items()]
    rows.sort(key=lambda r: (-r[3], r[0], r[1], r[2]))
    assert rows == [(r[0], r[1], int(r[2]), int(r[3])) for r in fx["q16"]]


def test_q17(fx, part):
    p = part
    sel = (p["brand"] == 23) & (p["container"] == 17)   # MED BOX
    pks = np.zeros(200001, bool)
    pks[p["partkey"][sel]] = True
    li = oracle.gen_lineitem(1.0, columns=["partkey", "quantity", "extendedprice"])
    m = pks[li["partkey"]]
    s = collections.defaultdict(lambda: [0.0, 0])
    for k, q in zip(li["partkey"][m].tolist(), li["quantity"][m].tolist()):
        s[k][0] += q
        s[k][1] += 1
    tot = 0.0
    for k, q, ep in zip(li["partkey"][m].tolist(), li["quantity"][m].tolist(),
                        li["extendedprice"][m].tolist()):
        if q < 0.2 * (s[k][0] / s[k][1]):
            tot += ep
    assert abs(tot / 7.0 - float(fx["q17"][0][0])) < 1e-4


def test_q19(fx, part):
    p = part
    li = oracle.gen_lineitem(1.0, columns=["orderkey", "partkey", "quantity",
                                           "extendedprice", "discount",
                                           "shipmode"])
    # shipinstruct stream aligned by per-order line counts
    M, A = 2147483647, 16807
    _, counts = np.unique(li["orderkey"], return_counts=True)
    sins = np.empty(len(li["orderkey"]), np.uint8)
    s = 1371272478
    i = 0
    for c in counts:
        for j in range(7):
            s = (s * A) % M
            if j < c:
                sins[i] = int(s / 2147483647.0 * 4)
                i += 1
    brand = np.zeros(200001, np.uint8); brand[p["partkey"]] = p["brand"]
    cont = np.zeros(200001, np.uint8); cont[p["partkey"]] = p["container"]
    size = np.zeros(200001, np.int32); size[p["partkey"]] = p["size"]
    b = brand[li["partkey"]]; c = cont[li["partkey"]]; sz = size[li["partkey"]]
    q = li["quantity"]
    base = (li["shipmode"] == 1) & (sins == 0)   # AIR, DELIVER IN PERSON
    m = (base & (b == 12) & np.isin(c, [0, 1, 4, 5]) & (q >= 1) & (q <= 11) & (sz >= 1) & (sz <= 5)) | \
        (base & (b == 23) & np.isin(c, [17, 18, 20, 21]) & (q >= 10) & (q <= 20) & (sz >= 1) & (sz <= 10)) | \
        (base & (b == 34) & np.isin(c, [8, 9, 12, 13]) & (q >= 20) & (q <= 30) & (sz >= 1) & (sz <= 15))
    rev = float(np.sum(li["extendedprice"][m] * (1.0 - li["discount"][m])))
    assert abs(rev - float(fx["q19"][0][0])) < 1e-3


def test_q20(fx, part, partsupp, supplier, strings_lib):
    p, ps, su = part, partsupp, supplier
    pkf = np.zeros(200001, bool)
    pkf[p["partkey"][p["name_ids"][:, 0] == oracle.COLOR_FOREST]] = True
    li = oracle.gen_lineitem(1.0, columns=["partkey", "suppkey", "quantity",
                                           "shipdate"])
    lm = (li["shipdate"] >= D(1994, 1, 1)) & (li["shipdate"] < D(1995, 1, 1)) \
        & pkf[li["partkey"]]
    qty = collections.defaultdict(float)
    for k, s, q in zip(li["partkey"][lm].tolist(), li["suppkey"][lm].tolist(),
                       li["quantity"][lm].tolist()):
        qty[(k, s)] += q
    nk = dict(zip(su["suppkey"].tolist(), su["nationkey"].tolist()))
    good = set()
    psm = pkf[ps["partkey"]]
    for k, s, aq in zip(ps["partkey"][psm].tolist(), ps["suppkey"][psm].tolist(),
                        ps["availqty"][psm].tolist()):
        # correlated sum is NULL (row excluded) with no 1994 shipments
        if (k, s) in qty and aq > 0.5 * qty[(k, s)]:
            good.add(s)
    rows = sorted(s for s in good if nk[s] == 3)   # CANADA
    exp = fx["q20"]
    assert len(rows) == len(exp)
    st = supplier_strings(strings_lib, rows, want=("name", "address"))
    for i, e in enumerate(exp):
        assert st["name"][i] == e[0] and st["address"][i] == e[1]


def test_q21(fx, orders, supplier):
    o, su = orders, supplier
    statF = set(o["orderkey"][o["orderstatus"] == 0].tolist())
    li = oracle.gen_lineitem(1.0, columns=["orderkey", "suppkey", "commitdate",
                                           "receiptdate"])
    late = li["receiptdate"] > li["commitdate"]
    u, starts = np.unique(li["orderkey"], return_index=True)
    skv = li["suppkey"]
    minA = np.minimum.reduceat(skv, starts)
    maxA = np.maximum.reduceat(skv, starts)
    minL = np.minimum.reduceat(np.where(late, skv, np.int64(1 << 40)), starts)
    maxL = np.maximum.reduceat(np.where(late, skv, np.int64(-1)), starts)
    nL = np.add.reduceat(late.astype(np.int64), starts)
    nk = np.zeros(10001, np.int32)
    nk[su["suppkey"]] = su["nationkey"]
    qual = (nL > 0) & (minA != maxA) & (minL == maxL) & np.isin(u, list(statF))
    cnt = collections.defaultdict(int)
    for s, c in zip(minL[qual].tolist(), nL[qual].tolist()):
        if nk[s] == 20:                      # SAUDI ARABIA
            cnt[s] += c
    got = [(f"Supplier#{k:09d}", v) for k, v in
           sorted(cnt.items(), key=lambda kv: (-kv[1], kv[0]))[:100]]
    assert got == [(r[0], int(r[1])) for r in fx["q21"]]


# ---- round-1 device-pinned queries, CPU-composed here as well so the whole
# 22-query semantic set is verifiable without a GPU (q3/q4/q6/q14/q15/q18;
# fixtures: hive_tpch qNN.result) ----

@pytest.fixture(scope="module")
def li():
    return oracle.gen_lineitem(1.0)


@pytest.fixture(scope="module")
def customer():
    return oracle.gen_customer(1.0)


def _rev_e4(l, sel):
    """exact revenue units of 1e-4: extprice_cents * (100 - disc_pct)"""
    cents = np.rint(l["extendedprice"][sel] * 100).astype(np.int64)
    pct = np.rint(l["discount"][sel] * 100).astype(np.int64)
    return cents * (100 - pct)


def _datestr(days):
    return str(datetime.date(1970, 1, 1) + datetime.timedelta(days=int(days)))


def test_q03(fx, li, orders, customer):
    cutoff = D(1995, 3, 15)
    bld = customer["custkey"][customer["mktsegment"] == 1]
    omask = (orders["orderdate"] < cutoff) & np.isin(orders["custkey"],
                                                     bld, kind="table")
    okeys = orders["orderkey"][omask]
    odate = dict(zip(okeys.tolist(), orders["orderdate"][omask].tolist()))
    lsel = (li["shipdate"] > cutoff) & np.isin(li["orderkey"], okeys,
                                               kind="table")
    rev = collections.defaultdict(int)
    for k, r in zip(li["orderkey"][lsel].tolist(), _rev_e4(li, lsel).tolist()):
        rev[k] += r
    rows = sorted(rev.items(), key=lambda kv: (-kv[1], odate[kv[0]]))[:10]
    exp = fx["q03"]
    assert len(rows) == len(exp)
    for (k, r), e in zip(rows, exp):
        assert k == int(e[0])
        assert abs(r / 1e4 - float(e[1])) < 5e-5
        assert _datestr(odate[k]) == e[2] and int(e[3]) == 0


def test_q04(fx, li, orders):
    late = np.unique(li["orderkey"][li["commitdate"] < li["receiptdate"]])
    omask = ((orders["orderdate"] >= D(1993, 7, 1)) &
             (orders["orderdate"] < D(1993, 10, 1)) &
             np.isin(orders["orderkey"], late, kind="table"))
    counts = np.bincount(orders["orderpriority"][omask], minlength=5)
    P = ["1-URGENT", "2-HIGH", "3-MEDIUM", "4-NOT SPECIFIED", "5-LOW"]
    exp = fx["q04"]
    assert len(exp) == 5
    for (name, n), e in zip(zip(P, counts.tolist()), exp):
        assert name == e[0] and n == int(e[1])


def test_q06(fx, li):
    sel = ((li["shipdate"] >= D(1994, 1, 1)) & (li["shipdate"] < D(1995, 1, 1)) &
           (li["discount"] >= 0.05 - 1e-9) & (li["discount"] <= 0.07 + 1e-9) &
           (li["quantity"] < 24))
    cents = np.rint(li["extendedprice"][sel] * 100).astype(np.int64)
    pct = np.rint(li["discount"][sel] * 100).astype(np.int64)
    total = int((cents * pct).sum())     # exact, units 1e-4
    assert abs(total / 1e4 - float(fx["q06"][0][0])) < 5e-5


def test_q14(fx, li, part):
    sel = (li["shipdate"] >= D(1995, 9, 1)) & (li["shipdate"] < D(1995, 10, 1))
    rev = _rev_e4(li, sel)
    tid = dict(zip(part["partkey"].tolist(), part["type_id"].tolist()))
    promo = np.array([tid[k] // 25 == 5 for k in li["partkey"][sel].tolist()])
    ratio = 100.0 * float(rev[promo].sum()) / float(rev.sum())
    assert abs(ratio - float(fx["q14"][0][0])) < 5e-7


def test_q15(fx, li, supplier):
    sel = (li["shipdate"] >= D(1996, 1, 1)) & (li["shipdate"] < D(1996, 4, 1))
    rev = collections.defaultdict(int)
    for s, r in zip(li["suppkey"][sel].tolist(), _rev_e4(li, sel).tolist()):
        rev[s] += r
    best = max(rev.values())
    tops = sorted(k for k, v in rev.items() if v == best)
    e = fx["q15"][0]
    assert len(tops) == len(fx["q15"]) == 1
    assert tops[0] == int(e[0]) and f"Supplier#{tops[0]:09d}" == e[1]
    assert abs(best / 1e4 - float(e[4])) < 5e-5


def test_q18(fx, li, orders):
    qty = collections.defaultdict(float)
    for k, q_ in zip(li["orderkey"].tolist(), li["quantity"].tolist()):
        qty[k] += q_
    big = {k: v for k, v in qty.items() if v > 300}
    info = {}
    for k, c, d, tp in zip(orders["orderkey"].tolist(),
                           orders["custkey"].tolist(),
                           orders["orderdate"].tolist(),
                           orders["totalprice_cents"].tolist()):
        if k in big:
            info[k] = (c, d, tp)
    rows = sorted(big.items(), key=lambda kv: (-info[kv[0]][2],
                                               info[kv[0]][1]))[:100]
    exp = fx["q18"]
    assert len(rows) == len(exp)
    for (k, sq), e in zip(rows, exp):
        c, d, tp = info[k]
        assert f"Customer#{c:09d}" == e[0] and c == int(e[1]) and k == int(e[2])
        assert _datestr(d) == e[3]
        assert abs(tp / 100.0 - float(e[4])) < 5e-3
        assert int(sq) == int(e[5])
