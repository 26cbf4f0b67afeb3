"""GPU parity tests for the operator-level C-ABI (filter/project, group-by
aggregation, hash join, page partitioner, row hash) against the oracle.

Tolerances (stated per test):
 - hashes, selection vectors, group keys/ids, counts, integer sums, partition
   assignment, join match sets: bit-exact;
 - f64 SUM/AVG through the generic aggregation operator: relative 1e-12
   (device atomicAdd order, like the reference's own cross-driver combine);
   the flagship Q1 path is exact (test_gpu_q1.py).
"""
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


def rng(seed=0):
    return np.random.default_rng(seed)


class TestHashRows:
    def test_bit_exact_multi_type(self, sess, ops):
        r = rng(1)
        n = 10000
        c1 = r.integers(-2**60, 2**60, n).astype(np.int64)
        c2 = r.standard_normal(n)
        c2[::100] = -0.0
        c3 = r.integers(-2**30, 2**30, n).astype(np.int32)
        page = ops.page_from_numpy([c1, c2, c3])
        got = ops.hash_rows(sess, page, [0, 1, 2])
        exp = oracle.hash_rows([c1, c2, c3], [oracle.TG_BIGINT, oracle.TG_DOUBLE, oracle.TG_INTEGER])
        assert np.array_equal(got, exp)


class TestFilter:
    def test_range_filter_selection_vector(self, sess, ops):
        r = rng(2)
        n = 50000
        sd = r.integers(8000, 11000, n).astype(np.int32)
        page = ops.page_from_numpy([sd])
        e = ops.expr(("col", 0), ("i64", 10471), "le")
        got = ops.filter_run(sess, e, page)
        exp = np.nonzero(sd <= 10471)[0].astype(np.int32)
        assert np.array_equal(got, exp)

    def test_and_or_between(self, sess, ops):
        r = rng(3)
        n = 20000
        a = r.uniform(0, 1, n)
        b = r.integers(1, 51, n).astype(np.int64)
        page = ops.page_from_numpy([a, b])
        # Q6 shape: disc between .05 and .07 AND qty < 24
        e = ops.expr(("col", 0), ("f64", 0.05), ("f64", 0.07), "between",
                     ("col", 1), ("i64", 24), "lt", "and")
        got = ops.filter_run(sess, e, page)
        exp = np.nonzero((a >= 0.05) & (a <= 0.07) & (b < 24))[0].astype(np.int32)
        assert np.array_equal(got, exp)

    def test_null_rejects_row(self, sess, ops):
        """any NULL argument rejects the row (nullable ColumnarFilter spec)"""
        n = 256
        a = np.arange(n, dtype=np.int64)
        valid = np.full((n + 63) // 64, ~np.uint64(0), np.uint64)
        valid[0] = np.uint64(0xFFFFFFFFFFFFFFF0)  # rows 0..3 null
        page = ops.page_from_numpy([a], valids=[valid])
        e = ops.expr(("col", 0), ("i64", 1000), "le")
        got = ops.filter_run(sess, e, page)
        assert np.array_equal(got, np.arange(4, n, dtype=np.int32))

    def test_input_selection_list(self, sess, ops):
        n = 1000
        a = np.arange(n, dtype=np.int64)
        page = ops.page_from_numpy([a])
        sel_pos = np.arange(0, n, 3, dtype=np.int32)
        sel = ops.TgSelected()
        sel.is_list = 1
        sel.offset = 0
        sel.size = len(sel_pos)
        sel.positions = sel_pos.ctypes.data
        e = ops.expr(("col", 0), ("i64", 500), "lt")
        got = ops.filter_run(sess, e, page, input_sel=sel)
        exp = sel_pos[a[sel_pos] < 500]
        assert np.array_equal(got, exp)

    def test_bigint_exact_above_2p53(self, sess, ops):
        """typed i64 lane: BIGINT compares are exact above 2^53 (the f64 lane
        would collapse base and base+1 to the same double). Exercises both the
        single-cmp fast path and the FTerms AND path."""
        base = (1 << 60) + 1
        a = np.array([base, base + 1, base - 1, base, 2, -base], dtype=np.int64)
        page = ops.page_from_numpy([a])
        got = ops.filter_run(sess, ops.expr(("col", 0), ("i64", base), "eq"), page)
        assert np.array_equal(got, np.nonzero(a == base)[0].astype(np.int32))
        got = ops.filter_run(sess, ops.expr(("col", 0), ("i64", base), "le"), page)
        assert np.array_equal(got, np.nonzero(a <= base)[0].astype(np.int32))
        # AND-of-terms path (k_filter_terms): a == base AND a > 0
        e = ops.expr(("col", 0), ("i64", base), "eq",
                     ("col", 0), ("i64", 0), "gt", "and")
        got = ops.filter_run(sess, e, page)
        assert np.array_equal(got, np.nonzero((a == base) & (a > 0))[0].astype(np.int32))
        # interpreter path (arith forces generic): (a - 1) == base
        e = ops.expr(("col", 0), ("i64", 1), "sub", ("i64", base), "eq")
        got = ops.filter_run(sess, e, page)
        assert np.array_equal(got, np.nonzero((a - 1) == base)[0].astype(np.int32))

    def test_kleene_three_valued_logic(self, sess, ops):
        """AND/OR/NOT follow SQL Kleene 3VL (io.trino.sql.ir.Logical):
        NOT(NULL OR FALSE) is NULL (reject), NOT(NULL AND FALSE) is TRUE
        (select). The round-1 null-to-false coercion selected the first and
        rejected the second."""
        n = 8
        a = np.array([1, 1, 0, 0, 1, 0, 1, 0], dtype=np.int64)
        b = np.array([1, 0, 1, 0, 1, 1, 0, 0], dtype=np.int64)
        # a is NULL on rows 0..3
        valid = np.array([~np.uint64(0) << np.uint64(4)], dtype=np.uint64)
        page = ops.page_from_numpy([a, b], valids=[valid, None])
        av = [None, None, None, None, True, False, True, False]
        bv = [True, False, True, False, True, True, False, False]

        def k_or(x, y):
            if x is True or y is True: return True
            if x is None or y is None: return None
            return False

        def k_and(x, y):
            if x is False or y is False: return False
            if x is None or y is None: return None
            return True

        def k_not(x):
            return None if x is None else not x

        # NOT (a OR b)
        e = ops.expr(("col", 0), ("i64", 0), "ne", ("col", 1), ("i64", 0), "ne",
                     "or", "not")
        got = ops.filter_run(sess, e, page)
        exp = [i for i in range(n) if k_not(k_or(av[i], bv[i])) is True]
        assert got.tolist() == exp
        # NOT (a AND b)
        e = ops.expr(("col", 0), ("i64", 0), "ne", ("col", 1), ("i64", 0), "ne",
                     "and", "not")
        got = ops.filter_run(sess, e, page)
        exp = [i for i in range(n) if k_not(k_and(av[i], bv[i])) is True]
        assert got.tolist() == exp
        # plain OR: NULL OR TRUE selects (true dominates null)
        e = ops.expr(("col", 0), ("i64", 0), "ne", ("col", 1), ("i64", 0), "ne", "or")
        got = ops.filter_run(sess, e, page)
        exp = [i for i in range(n) if k_or(av[i], bv[i]) is True]
        assert got.tolist() == exp

    def test_bigint_projection_lane(self, sess, ops):
        """non-identity BIGINT projection keeps exact int64 (orderkey+1 shape
        above 2^53)"""
        base = (1 << 60) + 7
        a = np.array([base, base + 2, 5, -base], dtype=np.int64)
        sd = np.array([0, 1, 0, 1], dtype=np.int32)
        page = ops.page_from_numpy([a, sd])
        f = ops.expr(("col", 1), ("i64", 0), "ge")   # select all
        projs = [ops.expr(("col", 0), ("i64", 1), "add")]
        op = ops.filter_project(sess, f, projs, out_types=[ops.TG_BIGINT])
        op.add_input(page)
        pages = op.drain()
        op.close()
        got = pages[0][0]["values"]
        assert got.dtype == np.int64
        assert np.array_equal(got, a + 1)

    def test_filter_project_operator(self, sess, ops):
        r = rng(4)
        n = 30000
        sd = r.integers(8000, 11000, n).astype(np.int32)
        ep = (r.integers(90100, 209900, n) / 100.0)
        di = r.integers(0, 11, n) / 100.0
        page = ops.page_from_numpy([sd, ep, di])
        f = ops.expr(("col", 0), ("i64", 10471), "le")
        projs = [ops.expr(("col", 1)),                                  # identity
                 ops.expr(("col", 1), ("f64", 1.0), ("col", 2), "sub", "mul")]  # ep*(1-di)
        op = ops.filter_project(sess, f, projs)
        op.add_input(page)
        pages = op.drain()
        op.close()
        assert len(pages) == 1
        mask = sd <= 10471
        assert np.array_equal(pages[0][0]["values"], ep[mask])
        exp = ep[mask] * (1.0 - di[mask])
        assert np.array_equal(pages[0][1]["values"], exp)  # same IEEE ops -> bit-exact


class TestHashAggregation:
    def test_bigint_groupby_counts_exact(self, sess, ops):
        r = rng(5)
        n = 200000
        keys = r.integers(0, 5000, n).astype(np.int64)
        vals = r.integers(0, 1000, n).astype(np.int64)
        page = ops.page_from_numpy([keys, vals])
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_I64, 1)])
        op.add_input(page)
        out = op.drain()[0]
        op.close()
        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        # output groups in first-occurrence (reference id) order
        assert np.array_equal(out[0]["values"], by_gid)
        assert np.array_equal(out[1]["values"], oracle.grouped_count(gids, ng))
        exp_sum = np.zeros(ng, np.int64)
        np.add.at(exp_sum, gids, vals)
        assert np.array_equal(out[2]["values"], exp_sum)

    def test_f64_sum_avg_tolerance(self, sess, ops):
        r = rng(6)
        n = 100000
        keys = r.integers(0, 37, n).astype(np.int64)
        vals = r.uniform(900, 105000, n)
        page = ops.page_from_numpy([keys, vals])
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_SUM_F64, 1), (ops.AGG_AVG_F64, 1)])
        op.add_input(page)
        out = op.drain()[0]
        op.close()
        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        exp_sum = oracle.grouped_sum_f64(gids, vals, ng)
        cnt = oracle.grouped_count(gids, ng)
        np.testing.assert_allclose(out[1]["values"], exp_sum, rtol=1e-12)
        np.testing.assert_allclose(out[2]["values"], exp_sum / cnt, rtol=1e-12)

    def test_multi_channel_keys_and_multipage(self, sess, ops):
        r = rng(7)
        pages_np = []
        for k in range(3):
            n = 4096
            k1 = r.integers(0, 9, n).astype(np.int8)
            k2 = r.integers(0, 7, n).astype(np.int32)
            v = r.integers(0, 100, n).astype(np.int64)
            pages_np.append((k1, k2, v))
        op = ops.hash_aggregation(sess, [0, 1], [ops.TG_TINYINT, ops.TG_INTEGER],
                                  [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_I64, 2)])
        for k1, k2, v in pages_np:
            op.add_input(ops.page_from_numpy([k1, k2, v]))
        out = op.drain()[0]
        op.close()
        allk1 = np.concatenate([p[0] for p in pages_np])
        allk2 = np.concatenate([p[1] for p in pages_np])
        allv = np.concatenate([p[2] for p in pages_np])
        gids, ng, first = oracle.flat_groupby([allk1, allk2],
                                              [oracle.TG_TINYINT, oracle.TG_INTEGER])
        assert np.array_equal(out[0]["values"], allk1[first])
        assert np.array_equal(out[1]["values"], allk2[first])
        assert np.array_equal(out[2]["values"], oracle.grouped_count(gids, ng))
        exp_sum = np.zeros(ng, np.int64)
        np.add.at(exp_sum, gids, allv)
        assert np.array_equal(out[3]["values"], exp_sum)

    def test_null_key_group(self, sess, ops):
        n = 128
        keys = np.arange(n, dtype=np.int64) % 5
        valid = np.full((n + 63) // 64, ~np.uint64(0), np.uint64)
        valid[0] &= ~np.uint64(1 << 7)  # row 7 (key 2) null
        valid[1] &= ~np.uint64(1 << 1)  # row 65 (key 0) null
        page = ops.page_from_numpy([keys], valids=[valid])
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1)])
        op.add_input(page)
        out = op.drain()[0]
        op.close()
        vb = np.array([(valid[i // 64] >> np.uint64(i % 64)) & np.uint64(1) for i in range(n)], bool)
        gids, ng, by_gid, nullg = oracle.bigint_groupby(keys, valid)
        assert len(out[0]["values"]) == ng
        # null group key emits NULL
        keyvalid = out[0]["valid"]
        assert keyvalid is not None
        nullbit = (keyvalid[nullg // 64] >> np.uint64(nullg % 64)) & np.uint64(1)
        assert nullbit == 0
        assert np.array_equal(out[1]["values"], oracle.grouped_count(gids, ng))

    def test_partial_final_roundtrip(self, sess, ops):
        """PARTIAL avg state (count,sum) pair -> FINAL combine == SINGLE"""
        r = rng(8)
        n = 50000
        keys = r.integers(0, 91, n).astype(np.int64)
        vals = r.uniform(0, 100, n)
        half = n // 2
        partial_pages = []
        for sl in (slice(0, half), slice(half, n)):
            op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                      [(ops.AGG_COUNT_STAR, -1), (ops.AGG_AVG_F64, 1)],
                                      step=ops.STEP_PARTIAL)
            op.add_input(ops.page_from_numpy([keys[sl], vals[sl]]))
            partial_pages.append(op.drain()[0])
            op.close()
        fin = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                   [(ops.AGG_COUNT_STAR, 1), (ops.AGG_AVG_F64, 2)],
                                   step=ops.STEP_FINAL)
        for p in partial_pages:
            cols = [p[0]["values"], p[1]["values"], p[2]["values"], p[3]["values"]]
            fin.add_input(ops.page_from_numpy(cols))
        out = fin.drain()[0]
        fin.close()
        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        # key order: first occurrence across partials arrival order
        cnt_exp = oracle.grouped_count(gids, ng)
        got = dict(zip(out[0]["values"].tolist(), out[1]["values"].tolist()))
        exp = dict(zip(by_gid.tolist(), cnt_exp.tolist()))
        assert got == exp
        sums = oracle.grouped_sum_f64(gids, vals, ng)
        got_avg = dict(zip(out[0]["values"].tolist(), out[2]["values"].tolist()))
        for k, g in zip(by_gid.tolist(), range(ng)):
            assert abs(got_avg[k] - sums[g] / cnt_exp[g]) <= 1e-12 * abs(got_avg[k])


class TestJoin:
    def _run_join(self, sess, ops, bk, bvals, pk, pvals):
        bridge = ops.JoinBridge(sess)
        build = ops.hash_builder(sess, bridge,
                                 [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
        build.add_input(ops.page_from_numpy([bk, bvals]))
        build.drain()
        build.close()
        probe = ops.lookup_join(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT],
                                [0], [0, 1])
        probe.add_input(ops.page_from_numpy([pk, pvals]))
        pages = probe.drain()
        probe.close()
        bridge.close()
        return pages

    def test_unique_keys_exact(self, sess, ops):
        """unique build keys: single match per probe row, reference order"""
        r = rng(9)
        nb, np_ = 5000, 20000
        bk = np.arange(nb, dtype=np.int64) * 8 + 1   # sparse like orderkeys
        bv = r.integers(0, 10**6, nb).astype(np.int64)
        pk = r.choice(np.concatenate([bk, np.arange(10**7, 10**7 + 5000)]), np_).astype(np.int64)
        pv = np.arange(np_, dtype=np.int64)
        pages = self._run_join(sess, ops, bk, bv, pk, pv)
        out = pages[0]
        t = oracle.JoinTable(bk)
        op_, ob_ = t.probe(pk)
        assert np.array_equal(out[0]["values"], pk[op_])
        assert np.array_equal(out[1]["values"], pv[op_])
        assert np.array_equal(out[2]["values"], bv[ob_])

    def test_duplicate_keys_match_set(self, sess, ops):
        """duplicate build keys: match SET equals oracle (chain order within a
        key is not deterministic on the wide build — DESIGN.md §6)"""
        r = rng(10)
        bk = r.integers(0, 50, 2000).astype(np.int64)
        bv = np.arange(2000, dtype=np.int64)
        pk = r.integers(0, 60, 500).astype(np.int64)
        pv = np.arange(500, dtype=np.int64)
        pages = self._run_join(sess, ops, bk, bv, pk, pv)
        out = pages[0]
        t = oracle.JoinTable(bk)
        op_, ob_ = t.probe(pk, cap=10**6)
        got = sorted(zip(out[1]["values"].tolist(), out[2]["values"].tolist()))
        exp = sorted(zip(pv[op_].tolist(), bv[ob_].tolist()))
        assert got == exp

    def test_null_keys_never_match(self, sess, ops):
        bk = np.array([1, 2, 3], np.int64)
        bv = np.array([10, 20, 30], np.int64)
        bvalid = np.array([~np.uint64(0)], np.uint64)
        bvalid[0] &= ~np.uint64(2)  # build row 1 null
        pk = np.array([2, 1], np.int64)
        pv = np.array([0, 1], np.int64)
        bridge = ops.JoinBridge(sess)
        build = ops.hash_builder(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
        build.add_input(ops.page_from_numpy([bk, bv], valids=[bvalid, None]))
        build.drain()
        build.close()
        probe = ops.lookup_join(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [0, 1])
        probe.add_input(ops.page_from_numpy([pk, pv]))
        out = probe.drain()[0]
        probe.close()
        bridge.close()
        assert out[0]["values"].tolist() == [1]   # only key 1 matches
        assert out[2]["values"].tolist() == [10]

    def test_q3_join_sf001(self, sess, ops):
        """orders x lineitem on orderkey at sf0.01 vs oracle probe"""
        o = oracle.gen_orders(0.01, 1, 15000)
        li = oracle.gen_lineitem(0.01)
        bridge = ops.JoinBridge(sess)
        build = ops.hash_builder(sess, bridge,
                                 [ops.TG_BIGINT, ops.TG_INTEGER], [0], [1])
        build.add_input(ops.page_from_numpy([o["orderkey"], o["orderdate"]]))
        build.drain()
        build.close()
        probe = ops.lookup_join(sess, bridge,
                                [ops.TG_BIGINT, ops.TG_DOUBLE], [0], [0, 1])
        probe.add_input(ops.page_from_numpy([li["orderkey"], li["extendedprice"]]))
        out = probe.drain()[0]
        probe.close()
        bridge.close()
        t = oracle.JoinTable(o["orderkey"])
        op_, ob_ = t.probe(li["orderkey"], cap=7_000_000)
        assert len(out[0]["values"]) == len(op_) == len(li["orderkey"])  # every line matches
        assert np.array_equal(out[0]["values"], li["orderkey"][op_])
        assert np.array_equal(out[1]["values"], li["extendedprice"][op_])
        assert np.array_equal(out[2]["values"], o["orderdate"][ob_])


class TestPartitioner:
    def test_partition_assignment_and_order(self, sess, ops):
        r = rng(11)
        n = 100000
        keys = r.integers(-2**62, 2**62, n).astype(np.int64)
        vals = r.standard_normal(n)
        page = ops.page_from_numpy([keys, vals])
        nparts = 8
        op = ops.page_partitioner(sess, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], nparts)
        op.add_input(page)
        op.finish()
        hashes = oracle.hash_rows([keys], [oracle.TG_BIGINT])
        exp_pid = np.array([oracle.partition_remote(int(np.int64(h)), nparts) for h in hashes],
                           np.int32)
        for p in range(nparts):
            cols = ops.get_partition(sess, op, p)
            rows = np.nonzero(exp_pid == p)[0]
            if cols is None:
                assert len(rows) == 0
                continue
            assert np.array_equal(cols[0]["values"], keys[rows])   # order preserved
            assert np.array_equal(cols[1]["values"], vals[rows])
        op.close()


class TestTopN:
    def test_topn_matches_reference_order(self, sess, ops):
        r = rng(12)
        n = 30000
        a = r.integers(0, 10**6, n).astype(np.int64)
        b = r.standard_normal(n)
        op = ops.topn(sess, [ops.TG_BIGINT, ops.TG_DOUBLE], [1, 0], [1, 0], 25)
        # multiple pages
        for sl in (slice(0, 10000), slice(10000, n)):
            op.add_input(ops.page_from_numpy([a[sl], b[sl]]))
        out = op.drain()[0]
        op.close()
        order = np.lexsort((a, -b))[:25]
        assert np.array_equal(out[0]["values"], a[order])
        assert np.array_equal(out[1]["values"], b[order])

    def test_topn_limit_exceeds_rows(self, sess, ops):
        a = np.array([3, 1, 2], np.int64)
        op = ops.topn(sess, [ops.TG_BIGINT], [0], [0], 10)
        op.add_input(ops.page_from_numpy([a]))
        out = op.drain()[0]
        op.close()
        assert out[0]["values"].tolist() == [1, 2, 3]


class TestDynamicFilter:
    def test_join_key_range(self, sess, ops):
        bk = np.array([42, 7, 99, 13], np.int64)
        bv = np.zeros(4, np.int64)
        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
        b.add_input(ops.page_from_numpy([bk, bv]))
        b.drain()
        b.close()
        mn, mx, nr = ops.join_key_range(bridge)
        assert (mn, mx, nr) == (7, 99, 4)
        bridge.close()


class TestSemiJoin:
    def test_matched_channel(self, sess, ops):
        bk = np.array([5, 9, 11], np.int64)
        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT], [0], [])
        b.add_input(ops.page_from_numpy([bk]))
        b.drain()
        b.close()
        pk = np.array([9, 4, 11, 5, 6], np.int64)
        pvalid = np.array([~np.uint64(0)], np.uint64)
        pvalid[0] &= ~np.uint64(1 << 1)   # probe row 1 null
        op = ops.semi_join(sess, bridge, 0)
        op.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
        out = op.drain()[0]
        op.close()
        bridge.close()
        assert out[0]["values"].tolist() == pk.tolist()
        assert out[1]["values"].tolist() == [1, 0, 1, 1, 0]
        mv = out[1]["valid"]
        assert mv is not None and not (mv[0] >> np.uint64(1)) & np.uint64(1)


class TestEncodedInputBlocks:
    """DictionaryBlock / RunLengthEncodedBlock inputs decode to flat values on
    upload (DictionaryBlock.java:55-58; SURVEY §8 micro-semantics: aggregation
    loops need only the flat forms)."""

    def test_dictionary_block_input(self, sess, ops):
        import ctypes
        r = rng(20)
        n = 10000
        dict_vals = np.array([100, 200, 300, 400], np.int64)
        ids = r.integers(0, 4, n).astype(np.int32)
        dict_block = ops.TgBlock()
        dict_block.type = ops.TG_BIGINT
        dict_block.kind = 0
        dict_block.position_count = 4
        dict_block.on_device = 0
        dict_block.data = dict_vals.ctypes.data
        blocks = (ops.TgBlock * 1)()
        blocks[0].type = ops.TG_BIGINT
        blocks[0].kind = 1  # DICTIONARY
        blocks[0].position_count = n
        blocks[0].on_device = 0
        blocks[0].ids = ids.ctypes.data
        blocks[0].dictionary = ctypes.pointer(dict_block)
        page = ops.TgPage()
        page.channel_count = 1
        page.position_count = n
        page.blocks = blocks
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], [(ops.AGG_COUNT_STAR, -1)])
        op.add_input(page)
        out = op.drain()[0]
        op.close()
        flat = dict_vals[ids]
        gids, ng, by_gid, _ = oracle.bigint_groupby(flat)
        assert np.array_equal(out[0]["values"], by_gid)
        assert np.array_equal(out[1]["values"], oracle.grouped_count(gids, ng))

    def test_rle_block_input(self, sess, ops):
        import ctypes
        n = 5000
        val = np.array([7.5], np.float64)
        vblock = ops.TgBlock()
        vblock.type = ops.TG_DOUBLE
        vblock.kind = 0
        vblock.position_count = 1
        vblock.on_device = 0
        vblock.data = val.ctypes.data
        blocks = (ops.TgBlock * 1)()
        blocks[0].type = ops.TG_DOUBLE
        blocks[0].kind = 2  # RLE
        blocks[0].position_count = n
        blocks[0].on_device = 0
        blocks[0].dictionary = ctypes.pointer(vblock)
        page = ops.TgPage()
        page.channel_count = 1
        page.position_count = n
        page.blocks = blocks
        e = ops.expr(("col", 0), ("f64", 7.0), "gt")
        got = ops.filter_run(sess, e, page)
        assert np.array_equal(got, np.arange(n, dtype=np.int32))


class TestOperatorStateMachine:
    """Operator.java:18-50 contract violations surface as errors, not UB."""

    def test_add_input_after_finish(self, sess, ops):
        import trino_amd
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], [(ops.AGG_COUNT_STAR, -1)])
        page = ops.page_from_numpy([np.arange(4, dtype=np.int64)])
        op.add_input(page)
        op.finish()
        with pytest.raises(trino_amd.TrinoGpuError):
            op.add_input(page)
        op.close()

    def test_probe_before_build_errors(self, sess, ops):
        import trino_amd
        bridge = ops.JoinBridge(sess)
        probe = ops.lookup_join(sess, bridge, [ops.TG_BIGINT], [0], [0])
        with pytest.raises(trino_amd.TrinoGpuError):
            probe.add_input(ops.page_from_numpy([np.arange(4, dtype=np.int64)]))
        probe.close()
        bridge.close()

    def test_varchar_value_block_rejected(self, sess, ops):
        import trino_amd
        blocks = (ops.TgBlock * 1)()
        blocks[0].type = ops.TG_VARCHAR
        blocks[0].kind = 0
        blocks[0].position_count = 1
        page = ops.TgPage()
        page.channel_count = 1
        page.position_count = 1
        page.blocks = blocks
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], [(ops.AGG_COUNT_STAR, -1)])
        with pytest.raises(trino_amd.TrinoGpuError) as ei:
            op.add_input(page)
        assert "dictionary-encoded" in str(ei.value)
        op.close()

    def test_empty_page_through_pipeline(self, sess, ops):
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1)])
        op.add_input(ops.page_from_numpy([np.empty(0, np.int64)]))
        out = op.drain()[0]
        op.close()
        assert len(out[0]["values"]) == 0


class TestMultiKeyJoin:
    """Generic multi-channel join keys (DefaultPagesHash analog): compound
    (BIGINT, INTEGER) key, duplicates, vs a numpy composition. Pairs compared
    as sets; probe rows ascending."""

    def test_compound_key(self, sess, ops):
        r = np.random.default_rng(33)
        nb, m = 4000, 20000
        b1 = r.integers(0, 500, nb).astype(np.int64)
        b2 = r.integers(0, 7, nb).astype(np.int32)
        bv = np.arange(nb, dtype=np.int64)
        p1 = r.integers(0, 600, m).astype(np.int64)
        p2 = r.integers(0, 8, m).astype(np.int32)
        pv = np.arange(m, dtype=np.int64)
        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge,
                             [ops.TG_BIGINT, ops.TG_INTEGER, ops.TG_BIGINT],
                             [0, 1], [2])
        b.add_input(ops.page_from_numpy([b1, b2, bv]))
        b.drain()
        b.close()
        probe = ops.lookup_join(sess, bridge,
                                [ops.TG_BIGINT, ops.TG_INTEGER, ops.TG_BIGINT],
                                [0, 1], [2])
        probe.add_input(ops.page_from_numpy([p1, p2, pv]))
        out = probe.drain()[0]
        probe.close()
        bridge.close()
        # numpy reference join
        bmap = {}
        for i in range(nb):
            bmap.setdefault((int(b1[i]), int(b2[i])), []).append(int(bv[i]))
        exp = []
        for i in range(m):
            for v in bmap.get((int(p1[i]), int(p2[i])), []):
                exp.append((int(pv[i]), v))
        got = sorted(zip(out[0]["values"].tolist(), out[1]["values"].tolist()))
        assert got == sorted(exp)

    def test_double_key_with_nulls(self, sess, ops):
        bk = np.array([1.5, 2.5, -0.0], np.float64)
        bv = np.array([10, 20, 30], np.int64)
        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_DOUBLE, ops.TG_BIGINT], [0], [1])
        b.add_input(ops.page_from_numpy([bk, bv]))
        b.drain()
        b.close()
        pk = np.array([2.5, 0.0, 7.0, 1.5], np.float64)   # +0.0 matches -0.0
        pvalid = np.array([~np.uint64(0)], np.uint64)
        pvalid[0] &= ~np.uint64(1 << 3)   # last probe row null
        probe = ops.lookup_join(sess, bridge, [ops.TG_DOUBLE], [0], [0])
        probe.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
        out = probe.drain()[0]
        probe.close()
        bridge.close()
        assert out[0]["values"].tolist() == [2.5, 0.0]
        assert out[1]["values"].tolist() == [20, 30]


class TestVarcharHash:
    def test_varchar_row_hash_and_partition(self, sess, ops):
        """canonical VARCHAR hash (XxHash64 over utf8 bytes) on device ==
        oracle, including the 31*combine with a BIGINT channel; and
        partitioning BY a varchar channel."""
        words = [b"", b"a", b"BUILDING", b"x" * 40, b"AIR REG", b"\xc3\xa9clair"]
        r = rng(44)
        n = 5000
        pick = r.integers(0, len(words), n)
        offsets = np.zeros(n + 1, np.int32)
        for i in range(n):
            offsets[i + 1] = offsets[i] + len(words[pick[i]])
        data = np.frombuffer(b"".join(words[k] for k in pick.tolist()), np.uint8).copy()
        keys = r.integers(0, 1000, n).astype(np.int64)
        page = ops.page_with_varchar([keys, (data, offsets)])
        got = ops.hash_rows(sess, page, [0, 1])
        M = (1 << 64) - 1
        exp = np.empty(n, np.uint64)
        for i in range(n):
            h = oracle.combine_hash(0, np.int64(np.uint64(oracle.bigint_hash(int(keys[i])))).item())
            h = oracle.combine_hash(h, np.int64(np.uint64(oracle.xxhash64(words[pick[i]]))).item())
            exp[i] = np.uint64(h % (1 << 64))
        assert np.array_equal(got, exp)


class TestVarcharGroupBy:
    """True VARCHAR group keys (variable-width key store; survey hard part
    (c)): group ids in row order, key bytes round-trip, counts exact."""

    def _varchar(self, words, pick):
        offsets = np.zeros(len(pick) + 1, np.int32)
        for i, k in enumerate(pick):
            offsets[i + 1] = offsets[i] + len(words[k])
        data = np.frombuffer(b"".join(words[k] for k in pick), np.uint8).copy()
        return data, offsets

    def test_varchar_keys(self, sess, ops):
        words = [b"AUTOMOBILE", b"BUILDING", b"", b"x" * 50, b"FURNITURE"]
        r = rng(55)
        n = 20000
        pick = r.integers(0, len(words), n).tolist()
        data, offsets = self._varchar(words, pick)
        vals = r.integers(0, 100, n).astype(np.int64)
        page = ops.page_with_varchar([(data, offsets), vals])
        op = ops.hash_aggregation(sess, [0], [ops.TG_VARCHAR],
                                  [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_I64, 1)])
        op.add_input(page)
        out = op.drain()[0]
        op.close()
        gids, ng, first = oracle.flat_groupby([data], [oracle.TG_VARCHAR],
                                              offsets=[offsets])
        assert len(out[0]["values"]) == ng
        exp_keys = [words[pick[i]] for i in first.tolist()]
        assert out[0]["values"] == exp_keys
        assert np.array_equal(out[1]["values"], oracle.grouped_count(gids, ng))
        es = np.zeros(ng, np.int64)
        np.add.at(es, gids, vals)
        assert np.array_equal(out[2]["values"], es)

    def test_varchar_plus_bigint_multipage(self, sess, ops):
        words = [b"F", b"O", b"NO", b"OF"]   # prefixes that must not collide
        r = rng(56)
        allp, allk = [], []
        op = ops.hash_aggregation(sess, [0, 1], [ops.TG_VARCHAR, ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1)])
        for pg in range(3):
            n = 3000
            pick = r.integers(0, len(words), n).tolist()
            keys = r.integers(0, 5, n).astype(np.int64)
            data, offsets = self._varchar(words, pick)
            op.add_input(ops.page_with_varchar([(data, offsets), keys]))
            allp += pick
            allk.append(keys)
        out = op.drain()[0]
        op.close()
        keys = np.concatenate(allk)
        data, offsets = self._varchar(words, allp)
        gids, ng, first = oracle.flat_groupby([data, keys],
                                              [oracle.TG_VARCHAR, oracle.TG_BIGINT],
                                              offsets=[offsets, None])
        assert len(out[0]["values"]) == ng
        assert out[0]["values"] == [words[allp[i]] for i in first.tolist()]
        assert np.array_equal(out[1]["values"], keys[first])
        assert np.array_equal(out[2]["values"], oracle.grouped_count(gids, ng))


class TestVarcharJoinAndGather:
    """VARCHAR join keys (generic CSR path, VarcharType byte-equality) and
    variable-width gather (VariableWidthBlock.copyPositions analog) through
    join outputs and filter/project."""

    def _varchar(self, words, pick):
        offsets = np.zeros(len(pick) + 1, np.int32)
        for i, k in enumerate(pick):
            offsets[i + 1] = offsets[i] + len(words[k])
        data = np.frombuffer(b"".join(words[k] for k in pick), np.uint8).copy()
        return data, offsets

    def test_varchar_join_key(self, sess, ops):
        words = [b"F", b"O", b"NO", b"OF", b"", b"BUILDING", b"x" * 40]
        r = rng(60)
        nb, m = 5000, 20000
        bpick = r.integers(0, len(words), nb).tolist()
        bval = r.integers(0, 10**6, nb).astype(np.int64)
        ppick = r.integers(0, len(words), m).tolist()
        pval = r.integers(0, 10**6, m).astype(np.int64)

        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_VARCHAR, ops.TG_BIGINT], [0], [0, 1])
        # two pages: exercises the varchar concat + offset rebase
        bd1, bo1 = self._varchar(words, bpick[:3008])
        bd2, bo2 = self._varchar(words, bpick[3008:])
        b.add_input(ops.page_with_varchar([(bd1, bo1), bval[:3008]]))
        b.add_input(ops.page_with_varchar([(bd2, bo2), bval[3008:]]))
        b.drain()
        j = ops.lookup_join(sess, bridge, [ops.TG_VARCHAR, ops.TG_BIGINT], [0], [0, 1])
        pd, po = self._varchar(words, ppick)
        j.add_input(ops.page_with_varchar([(pd, po), pval]))
        jp, _ = j.get_output()
        b.close()
        j.close()
        bridge.close()

        got = sorted(zip(jp[0]["values"], jp[1]["values"].tolist(),
                         jp[2]["values"], jp[3]["values"].tolist()))
        by_key = {}
        for k in range(nb):
            by_key.setdefault(words[bpick[k]], []).append(k)
        exp = sorted((words[ppick[i]], int(pval[i]), words[bpick[k]], int(bval[k]))
                     for i in range(m) for k in by_key.get(words[ppick[i]], ()))
        assert got == exp

    def test_varchar_multikey_join(self, sess, ops):
        # (varchar, bigint) composite key: prefixes must not cross-match
        words = [b"A", b"AB", b"B", b"BA"]
        r = rng(61)
        nb, m = 2000, 8000
        bpick = r.integers(0, len(words), nb).tolist()
        bk2 = r.integers(0, 3, nb).astype(np.int64)
        ppick = r.integers(0, len(words), m).tolist()
        pk2 = r.integers(0, 3, m).astype(np.int64)
        bridge = ops.JoinBridge(sess)
        bd, bo = self._varchar(words, bpick)
        b = ops.hash_builder(sess, bridge, [ops.TG_VARCHAR, ops.TG_BIGINT], [0, 1], [1])
        b.add_input(ops.page_with_varchar([(bd, bo), bk2]))
        b.drain()
        j = ops.lookup_join(sess, bridge, [ops.TG_VARCHAR, ops.TG_BIGINT], [0, 1], [1])
        pd, po = self._varchar(words, ppick)
        j.add_input(ops.page_with_varchar([(pd, po), pk2]))
        jp, _ = j.get_output()
        b.close()
        j.close()
        bridge.close()
        got = sorted(zip(jp[0]["values"].tolist(), jp[1]["values"].tolist()))
        by_key = {}
        for k in range(nb):
            by_key.setdefault((words[bpick[k]], int(bk2[k])), []).append(k)
        exp = sorted((int(pk2[i]), int(bk2[k]))
                     for i in range(m)
                     for k in by_key.get((words[ppick[i]], int(pk2[i])), ()))
        assert got == exp

    def test_varchar_semi_join(self, sess, ops):
        words = [b"RAIL", b"AIR", b"MAIL", b"SHIP", b"TRUCK"]
        r = rng(62)
        build_pick = [0, 2]   # IN ('RAIL','MAIL')
        m = 10000
        ppick = r.integers(0, len(words), m).tolist()
        bridge = ops.JoinBridge(sess)
        bd, bo = self._varchar(words, build_pick)
        b = ops.hash_builder(sess, bridge, [ops.TG_VARCHAR], [0], [])
        b.add_input(ops.page_with_varchar([(bd, bo)]))
        b.drain()
        sj = ops.semi_join(sess, bridge, 0)
        pd, po = self._varchar(words, ppick)
        sj.add_input(ops.page_with_varchar([(pd, po)]))
        sp, _ = sj.get_output()
        b.close()
        sj.close()
        bridge.close()
        exp = np.array([1 if ppick[i] in build_pick else 0 for i in range(m)], np.int8)
        assert np.array_equal(sp[-1]["values"].astype(np.int8), exp)

    def test_varchar_filter_project(self, sess, ops):
        words = [b"AUTOMOBILE", b"", b"HOUSEHOLD", b"MACHINERY" * 3]
        r = rng(63)
        n = 30000
        pick = r.integers(0, len(words), n).tolist()
        keys = r.integers(0, 100, n).astype(np.int64)
        data, offsets = self._varchar(words, pick)
        page = ops.page_with_varchar([keys, (data, offsets)])
        f = ops.expr(("col", 0), ("i64", 50), "lt")
        fp = ops.filter_project(sess, f, [ops.expr(("col", 0)), ops.expr(("col", 1))],
                                [ops.TG_BIGINT, ops.TG_VARCHAR])
        fp.add_input(page)
        fp.finish()
        out, _ = fp.get_output()
        fp.close()
        mask = keys < 50
        assert np.array_equal(out[0]["values"], keys[mask])
        assert out[1]["values"] == [words[pick[i]] for i in np.nonzero(mask)[0]]


class TestTopNDevicePath:
    """n >= 65536 routes TopN through the device radix-sort path
    (ops_topn.hip emit_device); parity vs numpy lexsort."""

    def test_large_topn_composite(self, sess, ops):
        r = rng(70)
        n = 200_000
        rev = r.uniform(0, 10**6, n)           # effectively unique f64
        date = r.integers(8000, 10000, n).astype(np.int32)
        okey = np.arange(n, dtype=np.int64)
        page = ops.page_from_numpy([okey, date, rev])
        top = ops.topn(sess, [ops.TG_BIGINT, ops.TG_INTEGER, ops.TG_DOUBLE],
                       [2, 1], [1, 0], 25)    # revenue DESC, date ASC
        top.add_input(page)
        out = top.drain()[0]
        order = np.lexsort((date, -rev))[:25]
        assert np.array_equal(out[0]["values"], okey[order])
        assert np.array_equal(out[1]["values"], date[order])
        assert np.array_equal(out[2]["values"], rev[order])

    def test_large_topn_asc_int(self, sess, ops):
        r = rng(71)
        n = 100_000
        v = r.integers(-10**9, 10**9, n).astype(np.int64)
        page = ops.page_from_numpy([v])
        top = ops.topn(sess, [ops.TG_BIGINT], [0], [0], 40)
        top.add_input(page)
        out = top.drain()[0]
        assert np.array_equal(out[0]["values"], np.sort(v)[:40])


class TestExactFixedPointSum:
    """TG_AGG_SUM_F64_EXACT: order-independent 128-bit fixed-point sum —
    bit-equal to the oracle's exact leg and bit-stable across page splits
    (plain SUM_F64's atomicAdd order is not)."""

    def test_bit_exact_vs_oracle(self, sess, ops):
        r = rng(80)
        n = 300_000
        keys = r.integers(0, 400, n).astype(np.int64)
        # money-scaled values: v*2^9 integer (cents * 2^-2 grid like Q1 prices)
        vals = r.integers(90100, 209900, n) / 100.0
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_SUM_F64_EXACT, 1, 43)])
        op.add_input(ops.page_from_numpy([keys, vals]))
        out = op.drain()[0]
        op.close()
        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        exp = oracle.grouped_sum_f64_exact(gids, vals, ng, scale_pow=43)
        assert np.array_equal(out[1]["values"], exp)   # bit-equal

    def test_split_invariance(self, sess, ops):
        r = rng(81)
        n = 200_000
        keys = r.integers(0, 101, n).astype(np.int64)
        vals = r.integers(0, 11, n) / 100.0            # discounts: v*2^59 integer
        results = []
        for splits in ([n], [64, n - 64], [1024, 77056, n - 78080]):
            op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                      [(ops.AGG_SUM_F64_EXACT, 1, 59)])
            at = 0
            for sz in splits:
                op.add_input(ops.page_from_numpy([keys[at:at + sz], vals[at:at + sz]]))
                at += sz
            out = op.drain()[0]
            op.close()
            results.append((np.asarray(out[0]["values"]), np.asarray(out[1]["values"])))
        for k2, v2 in results[1:]:
            assert np.array_equal(results[0][0], k2)
            assert np.array_equal(results[0][1], v2)   # bit-stable

    def test_partial_final_combine(self, sess, ops):
        r = rng(82)
        n = 120_000
        keys = r.integers(0, 37, n).astype(np.int64)
        vals = r.integers(90100, 209900, n) / 100.0
        part = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                    [(ops.AGG_SUM_F64_EXACT, 1, 43)],
                                    step=ops.STEP_PARTIAL)
        part.add_input(ops.page_from_numpy([keys, vals]))
        pout = part.drain()[0]
        part.close()
        fin = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                   [(ops.AGG_SUM_F64_EXACT, 1, 43)],
                                   step=ops.STEP_FINAL)
        fin.add_input(ops.page_from_numpy([np.asarray(pout[0]["values"]),
                                           np.asarray(pout[1]["values"]),
                                           np.asarray(pout[2]["values"])]))
        out = fin.drain()[0]
        fin.close()
        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        exp = oracle.grouped_sum_f64_exact(gids, vals, ng, scale_pow=43)
        assert np.array_equal(out[0]["values"], by_gid)
        assert np.array_equal(out[1]["values"], exp)


class TestPartitionedProbe:
    """Partitioned single-pass probe + stable match sort (large tables):
    thresholds forced down via env so the path runs at test sizes. Match
    LIST order is compared (probe-row order, duplicates reverse-insertion) —
    exactly the classic two-pass path's contract."""

    def test_matches_and_order_vs_classic(self, sess, ops):
        import os
        r = rng(90)
        nb, m = 50_000, 200_000
        # ~25% duplicate build keys (1-4 copies)
        base = r.choice(np.arange(10**7), 30_000, replace=False)
        bk = np.concatenate([base, r.choice(base, nb - len(base))]).astype(np.int64)
        r.shuffle(bk)
        bv = np.arange(nb, dtype=np.int64)
        pk = r.choice(np.concatenate([base, r.integers(10**8, 10**9, 50_000)]),
                      m).astype(np.int64)

        def run_join():
            bridge = ops.JoinBridge(sess)
            b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
            b.add_input(ops.page_from_numpy([bk, bv]))
            b.drain()
            j = ops.lookup_join(sess, bridge, [ops.TG_BIGINT], [0], [0])
            j.add_input(ops.page_from_numpy([pk]))
            jp, _ = j.get_output()
            b.close()
            j.close()
            bridge.close()
            return (jp[0]["values"].copy(), jp[1]["values"].copy())

        os.environ["TG_JOIN_PART"] = "1"
        os.environ["TG_JOIN_PART_MIN_ROWS"] = "1000"
        os.environ["TG_JOIN_PART_MIN_BYTES"] = "1000"
        try:
            got_p = run_join()
            os.environ["TG_JOIN_PART"] = "0"
            got_c = run_join()
        finally:
            os.environ.pop("TG_JOIN_PART", None)
            os.environ.pop("TG_JOIN_PART_MIN_ROWS", None)
            os.environ.pop("TG_JOIN_PART_MIN_BYTES", None)
        assert np.array_equal(got_p[0], got_c[0])   # identical list ORDER
        assert np.array_equal(got_p[1], got_c[1])


class TestSemiJoinNullSemantics:
    """Three-valued IN semantics (HashSemiJoinOperator.java:180-201):
    probe NULL -> false when the build set is empty, else NULL;
    a miss against a set containing NULL -> NULL."""

    def test_null_in_build_and_probe(self, sess, ops):
        bk = np.array([10, 20, 30, 0], np.int64)
        bvalid = np.array([0b0111], np.uint64)       # build row 3 is NULL
        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT], [0], [])
        b.add_input(ops.page_from_numpy([bk], valids=[bvalid]))
        b.drain()
        sj = ops.semi_join(sess, bridge, 0)
        pk = np.array([10, 99, 0, 20], np.int64)
        pvalid = np.array([0b1011], np.uint64)       # probe row 2 is NULL
        sj.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
        out, _ = sj.get_output()
        b.close()
        sj.close()
        bridge.close()
        m = out[-1]
        vals = m["values"].astype(np.int8)
        valid = np.asarray(m["valid"])
        def bit(i):
            return (int(valid[i >> 6]) >> (i & 63)) & 1
        assert vals[0] == 1 and bit(0) == 1          # 10 IN set -> true
        assert bit(1) == 0                           # 99 miss, set has NULL -> NULL
        assert bit(2) == 0                           # NULL probe, set nonempty -> NULL
        assert vals[3] == 1 and bit(3) == 1          # 20 -> true

    def test_null_probe_empty_build(self, sess, ops):
        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT], [0], [])
        b.add_input(ops.page_from_numpy([np.empty(0, np.int64)]))
        b.drain()
        sj = ops.semi_join(sess, bridge, 0)
        pk = np.array([5, 0], np.int64)
        pvalid = np.array([0b01], np.uint64)
        sj.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
        out, _ = sj.get_output()
        b.close()
        sj.close()
        bridge.close()
        m = out[-1]
        vals = m["values"].astype(np.int8)
        valid = np.asarray(m["valid"])
        assert vals[0] == 0 and ((int(valid[0]) >> 0) & 1) == 1   # miss, no null -> false
        assert vals[1] == 0 and ((int(valid[0]) >> 1) & 1) == 1   # NULL probe, empty set -> false


class TestStreamingAggregation:
    """StreamingAggregationOperator analog (ops_streamagg.hip): clustered
    single-BIGINT-key input, group ids = run order, page-spanning runs
    continue. Parity vs the hash aggregation operator on the same input."""

    def test_matches_hash_agg_multi_page(self, sess, ops):
        r = rng(95)
        # clustered keys: sorted with duplicates, split at a run boundary AND
        # mid-run across three pages
        keys = np.sort(r.integers(0, 40_000, 120_000)).astype(np.int64)
        vals = r.integers(90100, 209900, len(keys)) / 100.0
        iv = r.integers(0, 1000, len(keys)).astype(np.int64)
        cuts = [0, 50_000, 50_001, len(keys)]   # second page = 1 row (mid-run)
        sa = ops.streaming_aggregation(sess, 0,
                                       [(ops.AGG_COUNT_STAR, -1),
                                        (ops.AGG_SUM_F64_EXACT, 1, 43),
                                        (ops.AGG_SUM_I64, 2),
                                        (ops.AGG_AVG_F64, 1)])
        ha = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1),
                                   (ops.AGG_SUM_F64_EXACT, 1, 43),
                                   (ops.AGG_SUM_I64, 2),
                                   (ops.AGG_AVG_F64, 1)])
        for lo, hi in zip(cuts[:-1], cuts[1:]):
            page = [keys[lo:hi], vals[lo:hi], iv[lo:hi]]
            sa.add_input(ops.page_from_numpy(page))
            ha.add_input(ops.page_from_numpy(page))
        so = sa.drain()[0]
        ho = ha.drain()[0]
        sa.close()
        ha.close()
        assert np.array_equal(so[0]["values"], ho[0]["values"])   # keys, run order
        assert np.array_equal(so[1]["values"], ho[1]["values"])   # counts
        assert np.array_equal(so[2]["values"], ho[2]["values"])   # exact sums: bit-equal
        assert np.array_equal(so[3]["values"], ho[3]["values"])   # int sums
        np.testing.assert_allclose(so[4]["values"], ho[4]["values"], rtol=1e-12)

    def test_single_run_and_single_rows(self, sess, ops):
        sa = ops.streaming_aggregation(sess, 0, [(ops.AGG_COUNT_STAR, -1)])
        sa.add_input(ops.page_from_numpy([np.full(100, 7, np.int64)]))
        sa.add_input(ops.page_from_numpy([np.full(1, 7, np.int64)]))
        sa.add_input(ops.page_from_numpy([np.array([8], np.int64)]))
        out = sa.drain()[0]
        sa.close()
        assert out[0]["values"].tolist() == [7, 8]
        assert out[1]["values"].tolist() == [101, 1]


class TestSetBuilder:
    """SetBuilderOperator analog (ChannelSet): dense-range bitmap semi-join
    source. Results must equal the positional-index semi join, including the
    three-valued null semantics."""

    def test_bitmap_matches_index_semi(self, sess, ops):
        r = rng(120)
        nb, m = 40_000, 150_000
        bk = r.integers(10**6, 10**6 + 500_000, nb).astype(np.int64)
        b_nulls = r.random(nb) < 0.02
        bvalid = np.full((nb + 63) // 64, ~np.uint64(0), np.uint64)
        for i in np.nonzero(b_nulls)[0]:
            bvalid[i >> 6] &= ~np.uint64(1 << (i & 63))
        pk = r.integers(10**6 - 1000, 10**6 + 501_000, m).astype(np.int64)
        p_nulls = r.random(m) < 0.02
        pvalid = np.full((m + 63) // 64, ~np.uint64(0), np.uint64)
        for i in np.nonzero(p_nulls)[0]:
            pvalid[i >> 6] &= ~np.uint64(1 << (i & 63))

        def run(use_set):
            bridge = ops.JoinBridge(sess)
            if use_set:
                b = ops.set_builder(sess, bridge, [ops.TG_BIGINT], 0)
            else:
                b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT], [0], [])
            b.add_input(ops.page_from_numpy([bk], valids=[bvalid]))
            b.drain()
            sj = ops.semi_join(sess, bridge, 0)
            sj.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
            out, _ = sj.get_output()
            b.close()
            sj.close()
            bridge.close()
            return (out[-1]["values"].astype(np.int8).copy(),
                    np.asarray(out[-1]["valid"]).copy())

        vb, valb = run(True)
        vi, vali = run(False)
        assert np.array_equal(vb, vi)
        assert np.array_equal(valb, vali)


class TestProbeOuterJoin:
    """Probe-outer (LEFT) lookup join: unmatched probe rows (including null
    keys) emit one row with a NULL build side (LookupJoinOperators
    probe-outer), in probe-row order."""

    def test_left_join_with_nulls(self, sess, ops):
        r = rng(130)
        nb, m = 5000, 40_000
        bk = r.choice(np.arange(10**6), nb, replace=False).astype(np.int64)
        bv = r.integers(0, 10**6, nb).astype(np.int64)
        pk = r.choice(np.concatenate([bk, r.integers(10**7, 10**8, 20_000)]),
                      m).astype(np.int64)
        p_nulls = r.random(m) < 0.03
        pvalid = np.full((m + 63) // 64, ~np.uint64(0), np.uint64)
        for i in np.nonzero(p_nulls)[0]:
            pvalid[i >> 6] &= ~np.uint64(1 << (i & 63))

        bridge = ops.JoinBridge(sess)
        b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
        b.add_input(ops.page_from_numpy([bk, bv]))
        b.drain()
        j = ops.lookup_join(sess, bridge, [ops.TG_BIGINT], [0], [0], join_type=1)
        j.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
        jp, _ = j.get_output()
        b.close()
        j.close()
        bridge.close()

        bmap = dict(zip(bk.tolist(), bv.tolist()))
        got_k = jp[0]["values"]
        got_v = jp[1]["values"]
        vvalid = np.asarray(jp[1]["valid"])
        assert len(got_k) == m                      # every probe row exactly once
        for i in range(m):
            bit = (int(vvalid[i >> 6]) >> (i & 63)) & 1
            if p_nulls[i] or int(pk[i]) not in bmap:
                assert bit == 0                     # NULL build side
            else:
                assert bit == 1 and int(got_v[i]) == bmap[int(pk[i])]
        # probe keys pass through in order (null rows carry their slot)
        keep = ~p_nulls
        assert np.array_equal(np.asarray(got_k)[keep], pk[keep])


class TestDynamicFilter2:
    def test_fused_bitmap_filter(self, sess, ops):
        """DynamicPageFilter analog: request_bitmap on the build bridge, scan
        with filter_project_df — selection equals static predicate AND key
        membership; join results unchanged vs the unfiltered path."""
        r = rng(11)
        n = 200_000
        build_keys = np.unique(r.integers(0, 50_000, 5_000)).astype(np.int64)
        probe_keys = r.integers(0, 50_000, n).astype(np.int64)
        vals = r.standard_normal(n)
        bridge = ops.JoinBridge(sess)
        ops.request_bitmap(bridge)
        b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT], [0], [])
        b.add_input(ops.page_from_numpy([build_keys]))
        b.drain()
        page = ops.page_from_numpy([probe_keys, vals])
        f = ops.filter_project_df(sess, ops.expr(("col", 1), ("f64", 0.0), "gt"),
                                  [ops.expr(("col", 0)), ops.expr(("col", 1))],
                                  [ops.TG_BIGINT, ops.TG_DOUBLE], bridge, 0)
        f.add_input(page)
        pages = f.drain()
        f.close()
        got_k = pages[0][0]["values"]
        mask = (vals > 0) & np.isin(probe_keys, build_keys)
        assert np.array_equal(got_k, probe_keys[mask])
        # df-only scan (no static predicate)
        f2 = ops.filter_project_df(sess, None, [ops.expr(("col", 0))],
                                   [ops.TG_BIGINT], bridge, 0)
        f2.add_input(page)
        pages2 = f2.drain()
        f2.close()
        assert np.array_equal(pages2[0][0]["values"],
                              probe_keys[np.isin(probe_keys, build_keys)])
        b.close()
        bridge.close()


class TestMemoryAccounting:
    def test_oom_clean_and_accounting(self, sess, ops):
        """lib/trino-memory-context analog: an impossible device allocation
        fails with a clean error (no crash, session stays usable) and
        tg_session_memory reports pool growth."""
        import ctypes
        from trino_amd import _lib, TrinoGpuError
        _lib.tg_session_memory.restype = ctypes.c_int
        _lib.tg_session_memory.argtypes = [ctypes.c_void_p] * 3
        _lib.tg_device_malloc.restype = ctypes.c_int
        _lib.tg_device_malloc.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                          ctypes.c_int64]
        tot0 = ctypes.c_int64()
        cach0 = ctypes.c_int64()
        _lib.tg_session_memory(sess._h, ctypes.byref(tot0), ctypes.byref(cach0))
        p = ctypes.c_void_p()
        rc = _lib.tg_device_malloc(sess._h, ctypes.byref(p), 1 << 61)
        assert rc != 0          # clean failure, not a crash
        err = _lib.tg_last_error().decode()
        assert err              # error string populated
        # session still works
        q = ctypes.c_void_p()
        assert _lib.tg_device_malloc(sess._h, ctypes.byref(q), 1 << 20) == 0
        tot1 = ctypes.c_int64()
        cach1 = ctypes.c_int64()
        _lib.tg_session_memory(sess._h, ctypes.byref(tot1), ctypes.byref(cach1))
        # the impossible request evicts cached buffers (largest-first) before
        # failing, so total may SHRINK by up to the cached amount — but every
        # live (non-cached) byte must survive, plus the 1 MB just allocated
        assert tot1.value >= tot0.value - cach0.value + (1 << 20)
        _lib.tg_device_free(sess._h, q)


class TestDictionaryAwareFilter:
    def test_dict_filter_verdict_per_entry(self, sess, ops):
        """DictionaryAwareColumnarFilter analog: a dictionary-encoded channel
        filters by evaluating the predicate once per dictionary entry and
        expanding verdict[id] — identical selection to the flat evaluation."""
        r = rng(13)
        n = 300_000
        dvals = np.arange(0, 5000, dtype=np.int64) * 7 + 3
        ids = r.integers(0, len(dvals), n).astype(np.int32)
        flat = dvals[ids]
        e = ops.expr(("col", 0), ("i64", 17_000), "gt")
        page_d = ops.page_with_dict([(dvals, ids)])
        got = ops.filter_run(sess, e, page_d)
        exp = np.nonzero(flat > 17_000)[0].astype(np.int32)
        assert np.array_equal(got, exp)
        # arithmetic single-column predicate takes the dict path too
        e2 = ops.expr(("col", 0), ("i64", 3), "sub", ("i64", 7), "div",
                      ("i64", 2500), "lt")
        got2 = ops.filter_run(sess, e2, page_d)
        exp2 = np.nonzero((flat - 3) // 7 < 2500)[0].astype(np.int32)
        assert np.array_equal(got2, exp2)

    def test_dict_filter_speed(self, sess, ops):
        """measured win: dict path reads 4 B/row of ids vs 8 B/row flat (plus
        skipping the decode pass) — expect the dict-encoded filter to be
        faster at equal selectivity on a dict-heavy column."""
        import time
        r = rng(14)
        n = 20_000_000
        dvals = np.arange(0, 64, dtype=np.int64)
        ids = r.integers(0, 64, n).astype(np.int32)
        flat = dvals[ids].copy()
        e = ops.expr(("col", 0), ("i64", 32), "lt")
        page_d = ops.page_with_dict([(dvals, ids)])
        page_f = ops.page_from_numpy([flat])
        # warmup both
        ops.filter_run(sess, e, page_d)
        ops.filter_run(sess, e, page_f)
        t0 = time.time()
        for _ in range(3):
            ops.filter_run(sess, e, page_d)
        t_dict = time.time() - t0
        t0 = time.time()
        for _ in range(3):
            ops.filter_run(sess, e, page_f)
        t_flat = time.time() - t0
        # includes HtoD upload either way (ids half the bytes of values)
        assert t_dict < t_flat, (t_dict, t_flat)


class TestDenseAggregation:
    def test_dense_count_matches_hash(self, sess, ops):
        r = rng(21)
        n = 500_000
        keys = r.integers(1, 40_000, n).astype(np.int64)
        page = ops.page_from_numpy([keys])
        d = ops.dense_aggregation(sess, 0, 1, 40_000, (ops.AGG_COUNT_STAR, -1))
        d.add_input(page)
        d.finish()
        pages = d.drain()
        d.close()
        gk = pages[0][0]["values"]
        gc = pages[0][1]["values"]
        import collections
        exp = collections.Counter(keys.tolist())
        assert len(gk) == len(exp)
        # key-ordered emission
        assert np.all(np.diff(gk) > 0)
        for k, c in zip(gk.tolist(), gc.tolist()):
            assert exp[k] == c


class TestMaskedAggregates:
    def test_streaming_masked_min_max_count(self, sess, ops):
        """masked aggregation (AggregationMask predicate form): one pass
        computes unmasked and col_a>col_b-masked aggregates side by side —
        the Q21 shape (late-line min/max/count per order)."""
        r = rng(31)
        runs = 4000
        reps = r.integers(1, 8, runs)
        keys = np.repeat(np.arange(runs, dtype=np.int64) * 2 + 1, reps)
        n = len(keys)
        sk = r.integers(1, 1000, n).astype(np.int64)
        a = r.integers(0, 100, n).astype(np.int32)
        b = r.integers(0, 100, n).astype(np.int32)
        page = ops.page_from_numpy([keys, sk, b, a])   # mask: col3 > col2
        agg = ops.streaming_aggregation(
            sess, 0,
            [(ops.AGG_MIN_I64, 1), (ops.AGG_MAX_I64, 1),
             (ops.AGG_MIN_I64, 1, 0, 3, 2), (ops.AGG_MAX_I64, 1, 0, 3, 2),
             (ops.AGG_COUNT_STAR, -1, 0, 3, 2)])
        agg.add_input(page)
        agg.finish()
        pages = agg.drain()
        agg.close()
        out = pages[0]
        starts = np.concatenate([[0], np.cumsum(reps)[:-1]]).astype(np.int64)
        late = a > b
        skL = np.where(late, sk, np.int64(2**62))
        skH = np.where(late, sk, np.int64(-2**62))
        assert np.array_equal(out[1]["values"], np.minimum.reduceat(sk, starts))
        assert np.array_equal(out[2]["values"], np.maximum.reduceat(sk, starts))
        # groups with no late rows keep the min/max identities
        expL = np.minimum.reduceat(skL, starts)
        expH = np.maximum.reduceat(skH, starts)
        nlate = np.add.reduceat(late.astype(np.int64), starts)
        gotL = out[3]["values"]
        gotH = out[4]["values"]
        gotN = out[5]["values"]
        assert np.array_equal(gotN, nlate)
        m = nlate > 0
        assert np.array_equal(gotL[m], expL[m])
        assert np.array_equal(gotH[m], expH[m])

    def test_hash_masked_sum(self, sess, ops):
        r = rng(32)
        n = 200_000
        keys = r.integers(0, 500, n).astype(np.int64)
        v = r.integers(0, 1000, n).astype(np.int64)
        a = r.integers(0, 10, n).astype(np.int32)
        b = r.integers(0, 10, n).astype(np.int32)
        page = ops.page_from_numpy([keys, v, a, b])
        agg = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                   [(ops.AGG_SUM_I64, 1),
                                    (ops.AGG_SUM_I64, 1, 0, 2, 3)])
        agg.add_input(page)
        agg.finish()
        pages = agg.drain()
        agg.close()
        out = pages[0]
        import collections
        e_all = collections.defaultdict(int)
        e_m = collections.defaultdict(int)
        for k, vv, aa, bb in zip(keys.tolist(), v.tolist(), a.tolist(), b.tolist()):
            e_all[k] += vv
            if aa > bb:
                e_m[k] += vv
        for k, s_all, s_m in zip(out[0]["values"].tolist(),
                                 out[1]["values"].tolist(),
                                 out[2]["values"].tolist()):
            assert e_all[k] == s_all and e_m[k] == s_m


class TestMarkDistinct:
    def test_first_occurrence_flags(self, sess, ops):
        """MarkDistinctOperator analog: BOOLEAN channel true exactly on each
        key's first occurrence, across pages (global row order)."""
        r = rng(41)
        n = 50_000
        k1 = r.integers(0, 3000, n).astype(np.int64)
        k2 = r.integers(0, 4, n).astype(np.int32)
        v = r.standard_normal(n)
        op = ops.mark_distinct(sess, [0, 1], [ops.TG_BIGINT, ops.TG_INTEGER])
        half = n // 2
        p1 = ops.page_from_numpy([k1[:half], k2[:half], v[:half]])
        p2 = ops.page_from_numpy([k1[half:], k2[half:], v[half:]])
        op.add_input(p1)
        op.add_input(p2)
        op.finish()
        pages = op.drain()
        op.close()
        flags = np.concatenate([np.asarray(p[3]["values"]) for p in pages])
        seen = set()
        exp = np.zeros(n, np.int8)
        for i, key in enumerate(zip(k1.tolist(), k2.tolist())):
            if key not in seen:
                seen.add(key)
                exp[i] = 1
        assert np.array_equal(flags, exp)
        # pass-through channels intact
        vals = np.concatenate([np.asarray(p[2]["values"]) for p in pages])
        assert np.array_equal(vals, v)


class TestAdaptivePartialAgg:
    """PartialAggregationController analog (adaptive partial aggregation):
    a high-unique-ratio partial flush flips the shared controller; later
    partial operators re-shape pages into partial-state layout with no
    hash-table work; FINAL over the mixed partial outputs is exact."""

    def test_disable_pass_through_and_final_parity(self, sess, ops):
        r = rng(90)
        n = 60_000
        half = n // 2
        # ~88% unique in each half: above the 0.8 reference threshold
        keys = r.integers(1000, 1000 + 4 * half, n).astype(np.int64)
        vals_f = r.uniform(0, 100, n)
        vals_i = r.integers(-1000, 1000, n).astype(np.int64)
        aggs_p = [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_F64, 1),
                  (ops.AGG_SUM_I64, 2), (ops.AGG_MIN_I64, 2),
                  (ops.AGG_MAX_I64, 2)]
        # 1.5x threshold (150 KB) well under the first flush's 720 KB, and
        # the 200x re-enable bound (30 MB) well above it — a realistic
        # mid-stream disable (with tiny max_partial_bytes one big flush
        # crosses BOTH bounds in a single onFlush, like the reference would)
        ctrl = ops.PartialAggController(max_partial_bytes=100_000, threshold=0.8)
        partial_pages = []
        op1 = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], aggs_p,
                                   step=ops.STEP_PARTIAL)
        ctrl.attach(op1)
        assert not ctrl.disabled
        op1.add_input(ops.page_from_numpy([keys[:half], vals_f[:half],
                                           vals_i[:half]]))
        partial_pages += op1.drain()
        op1.close()
        assert ctrl.disabled          # flush saw ratio ~0.88 > 0.8

        op2 = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], aggs_p,
                                   step=ops.STEP_PARTIAL)
        ctrl.attach(op2)
        op2.add_input(ops.page_from_numpy([keys[half:], vals_f[half:],
                                           vals_i[half:]]))
        outs2 = op2.drain()
        op2.close()
        # pass-through: one output row per input row, states are the
        # per-row partial forms
        assert len(outs2) == 1
        p2 = outs2[0]
        assert np.array_equal(p2[0]["values"], keys[half:])
        assert np.all(np.asarray(p2[1]["values"]) == 1)          # COUNT
        assert np.array_equal(p2[2]["values"], vals_f[half:])    # SUM_F64
        assert np.array_equal(p2[3]["values"], vals_i[half:])    # SUM_I64
        assert np.array_equal(p2[4]["values"], vals_i[half:])    # MIN
        assert np.array_equal(p2[5]["values"], vals_i[half:])    # MAX
        partial_pages += outs2

        aggs_f = [(ops.AGG_COUNT_STAR, 1), (ops.AGG_SUM_F64, 2),
                  (ops.AGG_SUM_I64, 3), (ops.AGG_MIN_I64, 4),
                  (ops.AGG_MAX_I64, 5)]
        fin = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], aggs_f,
                                   step=ops.STEP_FINAL)
        for p in partial_pages:
            fin.add_input(ops.page_from_numpy(
                [np.asarray(b["values"]) for b in p]))
        out = fin.drain()[0]
        fin.close()
        ctrl.close()

        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        exp_cnt = oracle.grouped_count(gids, ng)
        got_key = np.asarray(out[0]["values"])
        order = {k: i for i, k in enumerate(got_key.tolist())}
        perm = np.array([order[k] for k in by_gid.tolist()])
        assert np.array_equal(got_key[perm], by_gid)
        assert np.array_equal(np.asarray(out[1]["values"])[perm], exp_cnt)
        exp_si = np.zeros(ng, np.int64)
        np.add.at(exp_si, gids, vals_i)
        assert np.array_equal(np.asarray(out[3]["values"])[perm], exp_si)
        exp_min = np.full(ng, np.iinfo(np.int64).max)
        np.minimum.at(exp_min, gids, vals_i)
        exp_max = np.full(ng, np.iinfo(np.int64).min)
        np.maximum.at(exp_max, gids, vals_i)
        assert np.array_equal(np.asarray(out[4]["values"])[perm], exp_min)
        assert np.array_equal(np.asarray(out[5]["values"])[perm], exp_max)
        exp_sf = oracle.grouped_sum_f64(gids, vals_f, ng)
        got_sf = np.asarray(out[2]["values"])[perm]
        assert np.allclose(got_sf, exp_sf, rtol=1e-12, atol=1e-9)

    def test_pass_through_avg_and_exact_states(self, sess, ops):
        """Two-channel partial states (AVG count+sum, EXACT lo+hi) through
        the pass-through path combine exactly in FINAL."""
        r = rng(91)
        n = 40_000
        keys = r.integers(0, 97, n).astype(np.int64)
        vals = r.integers(90100, 209900, n) / 100.0    # v*2^43 integer
        aggs_p = [(ops.AGG_AVG_F64, 1), (ops.AGG_SUM_F64_EXACT, 1, 43)]
        ctrl = ops.PartialAggController(max_partial_bytes=1000, threshold=0.8)
        # force-disable via a synthetic flush (reference: another driver of
        # the plan node already flipped it)
        ctrl.on_flush(2000, 100, 100, True)
        assert ctrl.disabled
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT], aggs_p,
                                  step=ops.STEP_PARTIAL)
        ctrl.attach(op)
        op.add_input(ops.page_from_numpy([keys, vals]))
        pout = op.drain()[0]
        op.close()
        ctrl.close()
        # layout: key, avg_cnt, avg_sum, exact_lo, exact_hi
        assert np.array_equal(pout[0]["values"], keys)
        assert np.all(np.asarray(pout[1]["values"]) == 1)
        assert np.array_equal(pout[2]["values"], vals)
        fin = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                   [(ops.AGG_AVG_F64, 1),
                                    (ops.AGG_SUM_F64_EXACT, 3, 43)],
                                   step=ops.STEP_FINAL)
        fin.add_input(ops.page_from_numpy(
            [np.asarray(b["values"]) for b in pout]))
        out = fin.drain()[0]
        fin.close()
        gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
        exp = oracle.grouped_sum_f64_exact(gids, vals, ng, scale_pow=43)
        assert np.array_equal(out[0]["values"], by_gid)
        assert np.array_equal(np.asarray(out[2]["values"]), exp)  # bit-exact
        cnt = oracle.grouped_count(gids, ng)
        got_avg = np.asarray(out[1]["values"])
        assert np.allclose(got_avg, exp / cnt, rtol=1e-12)

    def test_unsupported_specs_rejected(self, sess, ops):
        import trino_amd
        ctrl = ops.PartialAggController()
        # SINGLE step: rejected
        op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1)])
        with pytest.raises(trino_amd.TrinoGpuError):
            ctrl.attach(op)
        op.close()
        # VARCHAR group key: rejected
        op = ops.hash_aggregation(sess, [0], [ops.TG_VARCHAR],
                                  [(ops.AGG_COUNT_STAR, -1)],
                                  step=ops.STEP_PARTIAL)
        with pytest.raises(trino_amd.TrinoGpuError):
            ctrl.attach(op)
        op.close()
        ctrl.close()


class TestDedupSort:
    """tg_dedup_i64 (radix sort + unique compaction) == np.unique, including
    the reduced-bit-width path used by Q16's packed keys."""

    def test_matches_np_unique(self, sess, ops):
        import ctypes
        from trino_amd import copy_dtoh
        from trino_amd.ops import _lib
        _lib.tg_device_malloc.restype = ctypes.c_int
        _lib.tg_device_malloc.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                          ctypes.c_int64]
        _lib.tg_device_free.restype = ctypes.c_int
        _lib.tg_device_free.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        r = rng(95)
        for n, hi, bits in ((1, 10, 64), (1000, 50, 64),
                            (200_000, 1 << 40, 52), (500_000, 300_000, 64)):
            keys = r.integers(0, hi, n).astype(np.int64)
            d_in = ctypes.c_void_p()
            d_out = ctypes.c_void_p()
            assert _lib.tg_device_malloc(sess._h, ctypes.byref(d_in), n * 8) == 0
            assert _lib.tg_device_malloc(sess._h, ctypes.byref(d_out), n * 8) == 0
            assert _lib.tg_copy_htod(sess._h, d_in, keys.ctypes.data, n * 8) == 0
            m = ops.dedup_i64(sess, d_in, n, d_out, bits=bits)
            exp = np.unique(keys)
            assert m == len(exp)
            got = np.empty(m, np.int64)
            copy_dtoh(sess, got, d_out)
            assert np.array_equal(got, exp)        # sorted ascending
            _lib.tg_device_free(sess._h, d_in)
            _lib.tg_device_free(sess._h, d_out)


class TestDenseExactSum:
    def test_dense_exact_matches_hash_exact(self, sess, ops):
        """Dense-range SUM_F64_EXACT (2-word direct-array state) is
        bit-equal to the hash-aggregation exact sum on the same input."""
        r = rng(97)
        n = 400_000
        keys = r.integers(1, 5001, n).astype(np.int64)
        vals = r.integers(90100, 209900, n) / 100.0    # v*2^43 integer
        d = ops.dense_aggregation(sess, 0, 1, 5000,
                                  (ops.AGG_SUM_F64_EXACT, 1, 43))
        d.add_input(ops.page_from_numpy([keys, vals]))
        dout = d.drain()[0]
        d.close()
        h = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                                 [(ops.AGG_SUM_F64_EXACT, 1, 43)])
        h.add_input(ops.page_from_numpy([keys, vals]))
        hout = h.drain()[0]
        h.close()
        dk = np.asarray(dout[0]["values"])
        dv = np.asarray(dout[1]["values"])
        hk = np.asarray(hout[0]["values"])
        hv = np.asarray(hout[1]["values"])
        assert len(dk) == len(hk)
        order = np.argsort(hk, kind="stable")
        assert np.array_equal(dk, hk[order])           # key-ascending emit
        assert np.array_equal(dv, hv[order])           # bit-equal sums
