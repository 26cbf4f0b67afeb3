"""Device generator parity for the round-2 extended tables + text kernels
vs the CPU oracle (bit-exact; both restate the pinned streams in
oracle/tpch_text.h — the oracle side is itself pinned against the
reference's sf0.01 dataset in test_tpch_text.py)."""
import ctypes

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def sess():
    import trino_amd
    s = trino_amd.Session(0)
    yield s
    s.close()


@pytest.fixture(scope="module")
def ops():
    from trino_amd import ops
    return ops


def dbuf(sess, nbytes):
    from trino_amd import tpch_queries as tq
    return tq._device_buffer(sess, nbytes)


def dfree(sess, p):
    from trino_amd import tpch_queries as tq
    tq._device_free(sess, p)


def dl(sess, arr, ptr):
    from trino_amd import copy_dtoh
    return copy_dtoh(sess, arr, ptr)


SF = 0.1


class TestExtendedTables:
    def test_part2(self, sess, ops):
        from trino_amd import _lib, _check
        n = int(200_000 * SF)
        bufs = [dbuf(sess, n * 8), dbuf(sess, n * 2), dbuf(sess, n),
                dbuf(sess, n * 4), dbuf(sess, n), dbuf(sess, n * 5),
                dbuf(sess, n * 8)]
        _check(_lib.tg_tpch_gen_part2(sess._h, SF, 1, n, *bufs))
        exp = oracle.gen_part2(SF)
        got = [dl(sess, np.empty(n, np.int64), bufs[0]),
               dl(sess, np.empty(n, np.int16), bufs[1]),
               dl(sess, np.empty(n, np.uint8), bufs[2]),
               dl(sess, np.empty(n, np.int32), bufs[3]),
               dl(sess, np.empty(n, np.uint8), bufs[4]),
               dl(sess, np.empty(n * 5, np.uint8), bufs[5]),
               dl(sess, np.empty(n, np.int64), bufs[6])]
        assert np.array_equal(got[0], exp["partkey"])
        assert np.array_equal(got[1], exp["type_id"])
        assert np.array_equal(got[2], exp["brand"])
        assert np.array_equal(got[3], exp["size"])
        assert np.array_equal(got[4], exp["container"])
        assert np.array_equal(got[5].reshape(n, 5), exp["name_ids"])
        assert np.array_equal(got[6], exp["retail_cents"])
        for b in bufs:
            dfree(sess, b)

    def test_partsupp(self, sess, ops):
        from trino_amd import _lib, _check
        nparts = int(200_000 * SF)
        n = nparts * 4
        bufs = [dbuf(sess, n * 8), dbuf(sess, n * 8), dbuf(sess, n * 4),
                dbuf(sess, n * 8)]
        _check(_lib.tg_tpch_gen_partsupp(sess._h, SF, 1, nparts, *bufs))
        exp = oracle.gen_partsupp(SF)
        assert np.array_equal(dl(sess, np.empty(n, np.int64), bufs[0]), exp["partkey"])
        assert np.array_equal(dl(sess, np.empty(n, np.int64), bufs[1]), exp["suppkey"])
        assert np.array_equal(dl(sess, np.empty(n, np.int32), bufs[2]), exp["availqty"])
        assert np.array_equal(dl(sess, np.empty(n, np.int64), bufs[3]), exp["supplycost_cents"])
        for b in bufs:
            dfree(sess, b)

    def test_supplier2_orders3(self, sess, ops):
        from trino_amd import _lib, _check
        ns = int(10_000 * SF)
        b = [dbuf(sess, ns * 8), dbuf(sess, ns), dbuf(sess, ns * 8)]
        _check(_lib.tg_tpch_gen_supplier2(sess._h, SF, 1, ns, *b))
        exp = oracle.gen_supplier2(SF)
        assert np.array_equal(dl(sess, np.empty(ns, np.uint8), b[1]), exp["nationkey"])
        assert np.array_equal(dl(sess, np.empty(ns, np.int64), b[2]), exp["acctbal_cents"])
        for x in b:
            dfree(sess, x)
        no = int(1_500_000 * SF)
        bo = [dbuf(sess, no * 8), dbuf(sess, no * 8), dbuf(sess, no * 4),
              dbuf(sess, no), dbuf(sess, no), dbuf(sess, no * 8),
              dbuf(sess, no * 8), dbuf(sess, no * 4)]
        _check(_lib.tg_tpch_gen_orders3(sess._h, SF, 1, no, *bo))
        eo = oracle.gen_orders3(SF)
        assert np.array_equal(dl(sess, np.empty(no, np.int64), bo[0]), eo["orderkey"])
        assert np.array_equal(dl(sess, np.empty(no, np.uint8), bo[4]), eo["orderstatus"])
        assert np.array_equal(dl(sess, np.empty(no, np.int64), bo[5]), eo["totalprice_cents"])
        assert np.array_equal(dl(sess, np.empty(no, np.int64), bo[6]), eo["cmnt_off"])
        assert np.array_equal(dl(sess, np.empty(no, np.int32), bo[7]), eo["cmnt_len"])
        for x in bo:
            dfree(sess, x)

    def test_lineitem_shipinstruct(self, sess, ops):
        li = sess.tpch_lineitem(SF, with_shipinstruct=True)
        n = li.row_count
        got = dl(sess, np.empty(n, np.uint8), li.shipinstruct)
        exp = oracle.gen_lineitem(SF, columns=["shipinstruct"])["shipinstruct"]
        assert np.array_equal(got, exp)
        sess.tpch_lineitem_free(li)


class TestTextKernels:
    def test_pool_like_flags(self, sess, ops):
        from trino_amd import _lib, _check
        no = int(1_500_000 * SF)
        bo = [None, None, None, None, None, None,
              dbuf(sess, no * 8), dbuf(sess, no * 4)]
        _check(_lib.tg_tpch_gen_orders3(sess._h, SF, 1, no, *bo))
        d_flags = dbuf(sess, no)
        ops.pool_like_flags(sess, bo[6], bo[7], no, "%special%requests%", d_flags)
        got = dl(sess, np.empty(no, np.uint8), d_flags)
        eo = oracle.gen_orders3(SF)
        pool = oracle.text_pool_bytes().tobytes()
        exp = np.zeros(no, np.uint8)
        for i in range(no):
            s = pool[eo["cmnt_off"][i]:eo["cmnt_off"][i] + eo["cmnt_len"][i]]
            j = s.find(b"special")
            exp[i] = 1 if (j >= 0 and s.find(b"requests", j + 7) >= 0) else 0
        assert np.array_equal(got, exp)
        for x in (bo[6], bo[7], d_flags):
            dfree(sess, x)

    def test_supplier_comments_like(self, sess, ops):
        from trino_amd import _lib, _check
        ns = int(10_000 * SF)
        d_off = ctypes.c_void_p()
        d_bytes = ctypes.c_void_p()
        _check(_lib.tg_tpch_gen_supplier_comments(
            sess._h, SF, 1, ns, ctypes.byref(d_off), ctypes.byref(d_bytes)))
        d_flags = dbuf(sess, ns)
        ops.varchar_like_flags(sess, d_bytes, d_off, ns,
                               "%Customer%Complaints%", d_flags)
        got = dl(sess, np.empty(ns, np.uint8), d_flags)
        cm = oracle.gen_supplier_comments(1, ns)
        exp = np.array([1 if ("Customer" in c and
                              "Complaints" in c[c.find("Customer"):]) else 0
                        for c in cm], np.uint8)
        assert np.array_equal(got, exp)
        # and the comment text itself matches the oracle
        offs = dl(sess, np.empty(ns + 1, np.int32), d_off)
        total = int(offs[-1])
        by = dl(sess, np.empty(total, np.uint8), d_bytes)
        for i in (0, 1, ns // 2, ns - 1):
            s = bytes(by[offs[i]:offs[i + 1]]).decode("latin1")
            assert s == cm[i], i
        dfree(sess, d_flags)


class TestMinMaxAgg:
    def test_hash_agg_min_max(self, sess, ops):
        r = np.random.default_rng(7)
        n = 100_000
        keys = r.integers(0, 1000, n).astype(np.int64)
        vals = r.integers(-2**62, 2**62, n).astype(np.int64)
        page = ops.page_from_numpy([keys, vals])
        agg = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                   [(ops.AGG_MIN_I64, 1), (ops.AGG_MAX_I64, 1)])
        agg.add_input(page)
        agg.finish()
        pages = agg.drain()
        agg.close()
        out = pages[0]
        gk = out[0]["values"]
        gmin = out[1]["values"]
        gmax = out[2]["values"]
        import collections
        emin = collections.defaultdict(lambda: 2**63)
        emax = collections.defaultdict(lambda: -2**63)
        for k, v in zip(keys.tolist(), vals.tolist()):
            emin[k] = min(emin[k], v)
            emax[k] = max(emax[k], v)
        for k, mn, mx in zip(gk.tolist(), gmin.tolist(), gmax.tolist()):
            assert mn == emin[k] and mx == emax[k]

    def test_streaming_agg_min_max(self, sess, ops):
        r = np.random.default_rng(8)
        runs = 5000
        reps = r.integers(1, 9, runs)
        keys = np.repeat(np.arange(runs, dtype=np.int64) * 3, reps)
        n = len(keys)
        vals = r.integers(-2**40, 2**40, n).astype(np.int64)
        page = ops.page_from_numpy([keys, vals])
        agg = ops.streaming_aggregation(sess, 0,
                                        [(ops.AGG_MIN_I64, 1),
                                         (ops.AGG_MAX_I64, 1),
                                         (ops.AGG_COUNT_STAR, -1)])
        agg.add_input(page)
        agg.finish()
        pages = agg.drain()
        agg.close()
        out = pages[0]
        starts = np.concatenate([[0], np.cumsum(reps)[:-1]]).astype(np.int64)
        emin = np.minimum.reduceat(vals, starts)
        emax = np.maximum.reduceat(vals, starts)
        assert np.array_equal(out[1]["values"], emin)
        assert np.array_equal(out[2]["values"], emax)
        assert np.array_equal(out[3]["values"], reps.astype(np.int64))


class TestIndexedPoolLike:
    def test_indexed_vs_byte_scan(self, sess, ops):
        """The indexed pool LIKE (segment-occurrence index over the 300 MiB
        pool + per-row binary searches) must agree bitwise with the per-row
        byte-scan kernel on the same slices (TG_LIKE_IDX forces each path)."""
        import os
        from trino_amd import _lib, _check
        no = int(1_500_000 * SF)
        bo = [None, None, None, None, None, None,
              dbuf(sess, no * 8), dbuf(sess, no * 4)]
        _check(_lib.tg_tpch_gen_orders3(sess._h, SF, 1, no, *bo))
        for pat in ("%special%requests%", "%furiousl%", "%ironic%the%"):
            outs = []
            for mode in ("0", "1", "2"):    # byte scan / index / sorted scan
                d = dbuf(sess, no)
                os.environ["TG_LIKE_IDX"] = mode
                ops.pool_like_flags(sess, bo[6], bo[7], no, pat, d)
                outs.append(dl(sess, np.empty(no, np.uint8), d))
                dfree(sess, d)
            del os.environ["TG_LIKE_IDX"]
            assert np.array_equal(outs[0], outs[1]), pat
            assert np.array_equal(outs[0], outs[2]), pat
            assert outs[0].sum() > 0              # pattern actually matches
        for x in (bo[6], bo[7]):
            dfree(sess, x)
