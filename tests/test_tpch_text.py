"""Generator string-stream pins vs the reference's own sf0.01 dataset.

Golden source: tests/golden/tiny_sf001.json.gz, extracted from the
reference's Delta Lake test resources (testing/trino-testing-resources/...
databricks73) — a capture of dbgen sf0.01 output whose scale-independent
streams are byte-identical to the canonical SF1 rows the reference ships in
plugin/trino-example-http example-data. Rows inside Delta-UPDATE-tainted key
ranges are skipped (see tools/extract_ref_tiny.py).

Everything here runs on CPU against oracle/liboracle.so (tpch_text.h
restatement). The text pool comparison covers 3.5 MB of pool content across
86k+ comment slices from all six comment streams — any drift in the grammar
tables, pick mechanics, or stream seeds fails loudly.
"""
import ctypes
import gzip
import json
import os

import numpy as np
import pytest

import oracle

HERE = os.path.dirname(__file__)


@pytest.fixture(scope="module")
def tiny():
    with gzip.open(os.path.join(HERE, "golden", "tiny_sf001.json.gz"), "rt") as f:
        return json.load(f)["tables"]


@pytest.fixture(scope="module")
def lib():
    lib = ctypes.CDLL(os.path.join(HERE, "..", "oracle", "liboracle.so"))
    lib.tpch_text_pool.restype = ctypes.c_void_p
    lib.tpch_text_slices.argtypes = [ctypes.c_int64] * 3 + [ctypes.c_int32] * 3 + [ctypes.c_void_p] * 2
    lib.tpch_gen_vstr.argtypes = [ctypes.c_int64] * 3 + [ctypes.c_int32] * 2 + [ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p]
    lib.tpch_gen_part_names.argtypes = [ctypes.c_int64] * 2 + [ctypes.c_void_p]
    return lib


@pytest.fixture(scope="module")
def pool(lib):
    return lib.tpch_text_pool()


def slices(lib, seed, count, per_value, usage, avg):
    offs = np.zeros(count * per_value, np.int64)
    lens = np.zeros(count * per_value, np.int32)
    lib.tpch_text_slices(seed, 1, count, per_value, usage, avg,
                         offs.ctypes.data, lens.ctypes.data)
    return offs, lens


def tainted_keys(t):
    out = set()
    for lo, hi in t.get("tainted", []):
        out.update(range(lo + 1, hi + 1))
    return out


def pool_text(pool, off, ln):
    return ctypes.string_at(pool + int(off), int(ln)).decode("latin1")


SEED = {"o_cmnt": 276090261, "c_cmnt": 1335826707, "p_cmnt": 804159733,
        "ps_cmnt": 1961692154, "s_cmnt": 1341315363, "l_cmnt": 1095462486}


class TestTextPool:
    def test_orders_comments(self, tiny, lib, pool):
        t = tiny["orders"]
        cm = t["columns"]["comment"]
        offs, lens = slices(lib, SEED["o_cmnt"], len(cm), 1, 2, 49)
        taint = tainted_keys(t)
        bad = sum(1 for i, c in enumerate(cm)
                  if t["columns"]["orderkey"][i] not in taint
                  and pool_text(pool, offs[i], lens[i]) != c)
        assert bad == 0

    def test_customer_comments(self, tiny, lib, pool):
        t = tiny["customer"]
        cm = t["columns"]["comment"]
        offs, lens = slices(lib, SEED["c_cmnt"], len(cm), 1, 2, 73)
        taint = tainted_keys(t)
        bad = sum(1 for i, c in enumerate(cm)
                  if t["columns"]["custkey"][i] not in taint
                  and pool_text(pool, offs[i], lens[i]) != c)
        assert bad == 0

    def test_part_comments(self, tiny, lib, pool):
        t = tiny["part"]
        cm = t["columns"]["comment"]
        offs, lens = slices(lib, SEED["p_cmnt"], len(cm), 1, 2, 14)
        taint = tainted_keys(t)
        bad = sum(1 for i, c in enumerate(cm)
                  if t["columns"]["partkey"][i] not in taint
                  and pool_text(pool, offs[i], lens[i]) != c)
        assert bad == 0

    def test_partsupp_comments_bridge_order(self, tiny, lib, pool):
        t = tiny["partsupp"]
        bykey = {}
        c = t["columns"]
        for i in range(len(c["partkey"])):
            bykey[(c["partkey"][i], c["suppkey"][i])] = c["comment"][i]
        offs, lens = slices(lib, SEED["ps_cmnt"], 2000, 4, 8, 124)
        taint = tainted_keys(t)
        bad = k = 0
        for p in range(1, 2001):
            for j in range(4):
                sk = (p + j * (25 + (p - 1) // 100)) % 100 + 1
                if p not in taint and \
                        pool_text(pool, offs[k], lens[k]) != bykey[(p, sk)]:
                    bad += 1
                k += 1
        assert bad == 0

    def test_supplier_comments(self, tiny, lib, pool):
        cm = tiny["supplier"]["columns"]["comment"]
        offs, lens = slices(lib, SEED["s_cmnt"], len(cm), 1, 2, 63)
        # 100 suppliers at sf0.01: no BBB rows expected (verified: none
        # contain "Customer")
        bad = sum(1 for i, c in enumerate(cm)
                  if pool_text(pool, offs[i], lens[i]) != c)
        assert bad == 0

    def test_lineitem_comments(self, tiny, lib, pool):
        t = tiny["lineitem"]
        c = t["columns"]
        g = oracle.gen_lineitem(0.01)
        import collections
        lc = collections.Counter(g["orderkey"].tolist())
        okeys = sorted(lc)
        offs, lens = slices(lib, SEED["l_cmnt"], 15000, 7, 14, 27)
        taint = tainted_keys(t)
        bad = li = 0
        for oi, key in enumerate(okeys):
            for j in range(lc[key]):
                if key not in taint and \
                        pool_text(pool, offs[oi * 7 + j], lens[oi * 7 + j]) != c["comment"][li]:
                    bad += 1
                li += 1
        assert bad == 0


class TestStringStreams:
    def test_addresses(self, tiny, lib):
        for tbl, key, seed in (("supplier", "suppkey", 706178559),
                               ("customer", "custkey", 881155353)):
            t = tiny[tbl]
            n = len(t["columns"][key])
            buf = np.zeros((n, 64), np.uint8)
            lens = np.zeros(n, np.int32)
            lib.tpch_gen_vstr(seed, 1, n, 9, 25, buf.ctypes.data, 64,
                              lens.ctypes.data)
            taint = tainted_keys(t)
            for i, addr in enumerate(t["columns"]["address"]):
                if t["columns"][key][i] in taint:
                    continue
                got = bytes(buf[i, :lens[i]]).decode("latin1")
                assert got == addr, (tbl, i)

    def test_phones_acctbal_nationkey(self, tiny):
        import gzip as _
        for tbl, seed_ph, seed_ab, seed_nk in (
                ("supplier", 884434366, 962338209, 110356601),
                ("customer", 1521138112, 298370230, 1489529863)):
            t = tiny[tbl]["columns"]
            key = "suppkey" if tbl == "supplier" else "custkey"
            taint = tainted_keys(tiny[tbl])
            M, A = 2147483647, 16807
            sp, sa, sn = seed_ph, seed_ab, seed_nk
            for i in range(len(t[key])):
                draws = []
                for _k in range(3):
                    sp = (sp * A) % M
                    draws.append(sp)
                sa = (sa * A) % M
                sn = (sn * A) % M
                if t[key][i] in taint:
                    continue
                nk = int(sn / 2147483647.0 * 25)
                assert nk == t["nationkey"][i]
                bal = -99999 + int(sa / 2147483647.0 * 1099999)
                assert abs(bal / 100.0 - t["acctbal"][i]) < 1e-9
                l1 = 100 + int(draws[0] / 2147483647.0 * 900)
                l2 = 100 + int(draws[1] / 2147483647.0 * 900)
                l3 = 1000 + int(draws[2] / 2147483647.0 * 9000)
                assert f"{10+nk:02d}-{l1:03d}-{l2:03d}-{l3:04d}" == t["phone"][i]

    def test_part_names(self, tiny, lib):
        t = tiny["part"]["columns"]
        n = len(t["partkey"])
        ids = np.zeros((n, 5), np.uint8)
        lib.tpch_gen_part_names(1, n, ids.ctypes.data)
        taint = tainted_keys(tiny["part"])
        for i, name in enumerate(t["name"]):
            if t["partkey"][i] in taint:
                continue
            got = " ".join(COLORS[c] for c in ids[i])
            assert got == name, i

    def test_part_scalars(self, tiny):
        t = tiny["part"]["columns"]
        taint = tainted_keys(tiny["part"])
        M, A = 2147483647, 16807
        T1 = ["STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO"]
        T2 = ["ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"]
        T3 = ["TIN", "NICKEL", "BRASS", "STEEL", "COPPER"]
        C1 = ["SM", "LG", "MED", "JUMBO", "WRAP"]
        C2 = ["CASE", "BOX", "BAG", "JAR", "PACK", "PKG", "CAN", "DRUM"]
        sm, sb, st, ss, sc = 1, 46831694, 1841581359, 1193163244, 727633698
        for i in range(len(t["partkey"])):
            sm = (sm * A) % M
            sb = (sb * A) % M
            st = (st * A) % M
            ss = (ss * A) % M
            sc = (sc * A) % M
            if t["partkey"][i] in taint:
                continue
            mf = 1 + int(sm / 2147483647.0 * 5)
            assert f"Manufacturer#{mf}" == t["mfgr"][i]
            br = mf * 10 + 1 + int(sb / 2147483647.0 * 5)
            assert f"Brand#{br}" == t["brand"][i]
            ty = int(st / 2147483647.0 * 150)  # 0..149
            w1, w2, w3 = ty // 25, (ty // 5) % 5, ty % 5
            assert f"{T1[w1]} {T2[w2]} {T3[w3]}" == t["type"][i]
            assert 1 + int(ss / 2147483647.0 * 50) == t["size"][i]
            co = int(sc / 2147483647.0 * 40)  # 0..39
            assert f"{C1[co // 8]} {C2[co % 8]}" == t["container"][i]
            k = t["partkey"][i]
            rp = (90000 + (k // 10) % 20001 + 100 * (k % 1000)) / 100.0
            assert abs(rp - t["retailprice"][i]) < 1e-9

    def test_partsupp_scalars(self, tiny):
        t = tiny["partsupp"]["columns"]
        bykey = {}
        for i in range(len(t["partkey"])):
            bykey[(t["partkey"][i], t["suppkey"][i])] = (
                t["availqty"][i], t["supplycost"][i])
        taint = tainted_keys(tiny["partsupp"])
        M, A = 2147483647, 16807
        sq, sc = 1671059989, 1051288424
        for p in range(1, 2001):
            for j in range(4):
                sq = (sq * A) % M
                sc = (sc * A) % M
                if p in taint:
                    continue
                sk = (p + j * (25 + (p - 1) // 100)) % 100 + 1
                aq, cost = bykey[(p, sk)]
                assert 1 + int(sq / 2147483647.0 * 9999) == aq
                assert abs((100 + int(sc / 2147483647.0 * 99901)) / 100.0 - cost) < 1e-9

    def test_shipinstruct(self, tiny):
        t = tiny["lineitem"]["columns"]
        import collections
        g = oracle.gen_lineitem(0.01)
        lc = collections.Counter(g["orderkey"].tolist())
        okeys = sorted(lc)
        INS = ["DELIVER IN PERSON", "COLLECT COD", "TAKE BACK RETURN", "NONE"]
        M, A = 2147483647, 16807
        s = 1371272478
        taint = tainted_keys(tiny["lineitem"])
        li = 0
        for key in okeys:
            for j in range(7):
                s = (s * A) % M
                if j < lc[key]:
                    v = 1 + int(s / 2147483647.0 * 4)
                    if key not in taint:
                        assert INS[v - 1] == t["shipinstruct"][li], (key, j)
                    li += 1
        assert li == len(t["shipinstruct"])

    def test_orderstatus_derived(self, tiny):
        t = tiny["orders"]["columns"]
        lt = tiny["lineitem"]["columns"]
        st = {}
        for i in range(len(lt["orderkey"])):
            st.setdefault(lt["orderkey"][i], []).append(lt["linestatus"][i])
        taint = tainted_keys(tiny["orders"]) | tainted_keys(tiny["lineitem"])
        for i, k in enumerate(t["orderkey"]):
            if k in taint:
                continue
            ls = st[k]
            exp = ("F" if all(x == "F" for x in ls)
                   else "O" if all(x == "O" for x in ls) else "P")
            assert exp == t["orderstatus"][i]

    def test_orders_clerk(self, tiny):
        t = tiny["orders"]["columns"]
        taint = tainted_keys(tiny["orders"])
        M, A = 2147483647, 16807
        s = 1171034773
        for i in range(len(t["orderkey"])):
            s = (s * A) % M
            if t["orderkey"][i] in taint:
                continue
            c = 1 + int(s / 2147483647.0 * 1000)
            assert f"Clerk#{c:09d}" == t["clerk"][i]


COLORS = ["almond", "antique", "aquamarine", "azure", "beige", "bisque",
          "black", "blanched", "blue", "blush", "brown", "burlywood",
          "burnished", "chartreuse", "chiffon", "chocolate", "coral",
          "cornflower", "cornsilk", "cream", "cyan", "dark", "deep", "dim",
          "dodger", "drab", "firebrick", "floral", "forest", "frosted",
          "gainsboro", "ghost", "goldenrod", "green", "grey", "honeydew",
          "hot", "indian", "ivory", "khaki", "lace", "lavender", "lawn",
          "lemon", "light", "lime", "linen", "magenta", "maroon", "medium",
          "metallic", "midnight", "mint", "misty", "moccasin", "navajo",
          "navy", "olive", "orange", "orchid", "pale", "papaya", "peach",
          "peru", "pink", "plum", "powder", "puff", "purple", "red", "rose",
          "rosy", "royal", "saddle", "salmon", "sandy", "seashell", "sienna",
          "sky", "slate", "smoke", "snow", "spring", "steel", "tan",
          "thistle", "tomato", "turquoise", "violet", "wheat", "white",
          "yellow"]
