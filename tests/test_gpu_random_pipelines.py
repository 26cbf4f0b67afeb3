"""Randomized operator-pipeline property tests: for seeded random inputs and
random plan shapes (filter -> [join] -> group-by), the operator chain must
match an independent numpy composition. Integer outputs bit-exact; f64 sums
rtol 1e-12 (generic-agg tolerance).
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


@pytest.mark.parametrize("seed", range(8))
def test_random_filter_groupby(sess, ops, seed):
    r = np.random.default_rng(seed)
    n = int(r.integers(1, 60_000))
    nkeys = int(r.integers(1, 200))
    keys = r.integers(0, nkeys, n).astype(np.int64)
    vals_f = r.uniform(-1000, 1000, n)
    vals_i = r.integers(-10**6, 10**6, n).astype(np.int64)
    thresh = int(r.integers(0, nkeys))
    page = ops.page_from_numpy([keys, vals_f, vals_i])

    f = ops.expr(("col", 0), ("i64", thresh), "ge")
    fp = ops.filter_project(sess, f,
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2))],
                            [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_BIGINT])
    fp.add_input(page)
    fp.finish()
    sel_page, fin = fp.get_output()
    # re-upload host copy into the agg (also exercises download path)
    agg = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                               [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_F64, 1),
                                (ops.AGG_SUM_I64, 2)])
    if sel_page is not None:
        agg.add_input(ops.page_from_numpy([sel_page[0]["values"],
                                           sel_page[1]["values"],
                                           sel_page[2]["values"]]))
    out = agg.drain()
    fp.close()
    agg.close()

    mask = keys >= thresh
    ek, ev, ei = keys[mask], vals_f[mask], vals_i[mask]
    if len(ek) == 0:
        assert not out or len(out[0][0]["values"]) == 0
        return
    gids, ng, by_gid, _ = oracle.bigint_groupby(ek)
    o = out[0]
    assert np.array_equal(o[0]["values"], by_gid)
    assert np.array_equal(o[1]["values"], oracle.grouped_count(gids, ng))
    es = np.zeros(ng)
    np.add.at(es, gids, ev)
    np.testing.assert_allclose(o[2]["values"], es, rtol=1e-12, atol=1e-9)
    eis = np.zeros(ng, np.int64)
    np.add.at(eis, gids, ei)
    assert np.array_equal(o[3]["values"], eis)


@pytest.mark.parametrize("seed", range(6))
def test_random_join_agg(sess, ops, seed):
    r = np.random.default_rng(100 + seed)
    nb = int(r.integers(1, 20_000))
    m = int(r.integers(1, 80_000))
    dup = int(r.integers(1, 4))
    bk = np.repeat(r.choice(np.arange(10**6), nb // dup + 1, replace=False), dup)[:nb].astype(np.int64)
    r.shuffle(bk)
    bv = r.integers(0, 10**6, nb).astype(np.int64)
    pk = r.choice(np.concatenate([bk, r.integers(10**7, 10**8, max(m // 3, 1))]), m).astype(np.int64)
    pv = r.uniform(0, 100, m)

    bridge = ops.JoinBridge(sess)
    b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
    b.add_input(ops.page_from_numpy([bk, bv]))
    b.drain()
    j = ops.lookup_join(sess, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], [0, 1])
    j.add_input(ops.page_from_numpy([pk, pv]))
    jp, _ = j.get_output()
    b.close()

    t = oracle.JoinTable(bk)
    op_, ob_ = t.probe(pk, cap=8 * m + 64)
    if jp is None:
        assert len(op_) == 0
        j.close()
        bridge.close()
        return
    got = sorted(zip(jp[0]["values"].tolist(), jp[1]["values"].tolist(),
                     jp[2]["values"].tolist()))
    exp = sorted(zip(pk[op_].tolist(), pv[op_].tolist(), bv[ob_].tolist()))
    assert got == exp
    j.close()
    bridge.close()


@pytest.mark.parametrize("seed", range(4))
def test_random_partition_roundtrip(sess, ops, seed):
    r = np.random.default_rng(200 + seed)
    n = int(r.integers(1, 120_000))
    nparts = int(r.choice([2, 3, 8, 17]))
    keys = r.integers(-2**62, 2**62, n).astype(np.int64)
    vals = r.standard_normal(n)
    op = ops.page_partitioner(sess, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], nparts)
    op.add_input(ops.page_from_numpy([keys, vals]))
    op.finish()
    hashes = oracle.hash_rows([keys], [oracle.TG_BIGINT])
    exp_pid = np.array([oracle.partition_remote(int(np.int64(h)), nparts) for h in hashes],
                       np.int32)
    seen = 0
    for p in range(nparts):
        cols = ops.get_partition(sess, op, p)
        rows = np.nonzero(exp_pid == p)[0]
        if cols is None:
            assert len(rows) == 0
            continue
        assert np.array_equal(cols[0]["values"], keys[rows])
        assert np.array_equal(cols[1]["values"], vals[rows])
        seen += len(rows)
    assert seen == n
    op.close()


@pytest.mark.parametrize("seed", range(4))
def test_random_semi_join_with_nulls(sess, ops, seed):
    """Semi join three-valued logic on random data with null keys on both
    sides (HashSemiJoinOperator.java:180-201)."""
    r = np.random.default_rng(300 + seed)
    nb = int(r.integers(1, 5000))
    m = int(r.integers(1, 40_000))
    bk = r.integers(0, 3000, nb).astype(np.int64)
    b_nulls = r.random(nb) < 0.05
    bvalid = np.full((nb + 63) // 64, ~np.uint64(0), np.uint64)
    for i in np.nonzero(b_nulls)[0]:
        bvalid[i >> 6] &= ~np.uint64(1 << (i & 63))
    pk = r.integers(0, 6000, m).astype(np.int64)
    p_nulls = r.random(m) < 0.05
    pvalid = np.full((m + 63) // 64, ~np.uint64(0), np.uint64)
    for i in np.nonzero(p_nulls)[0]:
        pvalid[i >> 6] &= ~np.uint64(1 << (i & 63))

    bridge = ops.JoinBridge(sess)
    b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT], [0], [])
    b.add_input(ops.page_from_numpy([bk], valids=[bvalid]))
    b.drain()
    sj = ops.semi_join(sess, bridge, 0)
    sj.add_input(ops.page_from_numpy([pk], valids=[pvalid]))
    out, _ = sj.get_output()
    b.close()
    sj.close()
    bridge.close()

    bset = set(bk[~b_nulls].tolist())
    build_has_null = bool(b_nulls.any())
    vals = out[-1]["values"].astype(np.int8)
    valid = np.asarray(out[-1]["valid"])
    for i in range(m):
        bit = (int(valid[i >> 6]) >> (i & 63)) & 1
        if p_nulls[i]:
            exp_null = nb > 0            # set nonempty -> NULL
            assert bit == (0 if exp_null else 1)
        elif int(pk[i]) in bset:
            assert vals[i] == 1 and bit == 1
        elif build_has_null:
            assert bit == 0              # miss vs set containing NULL -> NULL
        else:
            assert vals[i] == 0 and bit == 1


@pytest.mark.parametrize("seed", range(4))
def test_random_exact_sum_pipeline(sess, ops, seed):
    """filter -> group-by with TG_AGG_SUM_F64_EXACT: bit-equal to the
    oracle's exact leg on money-grid values, any page split."""
    r = np.random.default_rng(400 + seed)
    n = int(r.integers(1000, 150_000))
    keys = r.integers(0, 500, n).astype(np.int64)
    vals = r.integers(90100, 209900, n) / 100.0      # >= 2^9: 2^-43 grid
    thresh = int(r.integers(0, 500))
    split = int(r.integers(1, n))
    op = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                              [(ops.AGG_SUM_F64_EXACT, 1, 43)])
    f = ops.expr(("col", 0), ("i64", thresh), "lt")
    for lo, hi in ((0, split), (split, n)):
        if hi == lo:
            continue
        fp = ops.filter_project(sess, f,
                                [ops.expr(("col", 0)), ops.expr(("col", 1))],
                                [ops.TG_BIGINT, ops.TG_DOUBLE])
        fp.add_input(ops.page_from_numpy([keys[lo:hi], vals[lo:hi]]))
        fp.finish()
        sel, _ = fp.get_output()
        fp.close()
        if sel is not None:
            op.add_input(ops.page_from_numpy([sel[0]["values"], sel[1]["values"]]))
    out = op.drain()
    op.close()
    mask = keys < thresh
    ek, ev = keys[mask], vals[mask]
    if len(ek) == 0:
        assert not out or len(out[0][0]["values"]) == 0
        return
    gids, ng, by_gid, _ = oracle.bigint_groupby(ek)
    exp = oracle.grouped_sum_f64_exact(gids, ev, ng, scale_pow=43)
    o = out[0]
    assert np.array_equal(o[0]["values"], by_gid)
    assert np.array_equal(o[1]["values"], exp)


@pytest.mark.parametrize("seed", range(3))
def test_random_outer_join_multipage(sess, ops, seed):
    """Probe-outer join across multiple probe pages vs a dict composition."""
    r = np.random.default_rng(500 + seed)
    nb = int(r.integers(1, 8000))
    bk = r.choice(np.arange(10**6), nb, replace=False).astype(np.int64)
    bv = r.standard_normal(nb)
    bridge = ops.JoinBridge(sess)
    b = ops.hash_builder(sess, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], [1])
    b.add_input(ops.page_from_numpy([bk, bv]))
    b.drain()
    j = ops.lookup_join(sess, bridge, [ops.TG_BIGINT], [0], [0], join_type=1)
    bmap = dict(zip(bk.tolist(), bv.tolist()))
    outs = []
    for pg in range(3):
        m = int(r.integers(1, 20_000))
        pk = r.choice(np.concatenate([bk, r.integers(10**7, 10**8, 5000)]),
                      m).astype(np.int64)
        j.add_input(ops.page_from_numpy([pk]))
        jp, _ = j.get_output()
        assert len(jp[0]["values"]) == m
        vvalid = np.asarray(jp[1]["valid"])
        for i in range(m):
            bit = (int(vvalid[i >> 6]) >> (i & 63)) & 1
            if int(pk[i]) in bmap:
                assert bit == 1 and jp[1]["values"][i] == bmap[int(pk[i])]
            else:
                assert bit == 0
        outs.append(jp)
    j.close()
    b.close()
    bridge.close()


@pytest.mark.parametrize("seed", range(3))
def test_random_streaming_partial_final(sess, ops, seed):
    """Streaming PARTIAL -> hash FINAL over clustered keys: the distributed
    aggregation shape, exact sums bit-equal to the oracle."""
    r = np.random.default_rng(600 + seed)
    n = int(r.integers(1000, 80_000))
    keys = np.sort(r.integers(0, n // 3 + 1, n)).astype(np.int64)
    vals = r.integers(90100, 209900, n) / 100.0
    part = ops.streaming_aggregation(sess, 0,
                                     [(ops.AGG_COUNT_STAR, -1),
                                      (ops.AGG_SUM_F64_EXACT, 1, 43)],
                                     step=ops.STEP_PARTIAL)
    part.add_input(ops.page_from_numpy([keys, vals]))
    pout = part.drain()[0]
    part.close()
    fin = ops.hash_aggregation(sess, [0], [ops.TG_BIGINT],
                               [(ops.AGG_COUNT_STAR, 1),
                                (ops.AGG_SUM_F64_EXACT, 2, 43)],
                               step=ops.STEP_FINAL)
    fin.add_input(ops.page_from_numpy([np.asarray(pout[0]["values"]),
                                       np.asarray(pout[1]["values"]),
                                       np.asarray(pout[2]["values"]),
                                       np.asarray(pout[3]["values"])]))
    out = fin.drain()[0]
    fin.close()
    gids, ng, by_gid, _ = oracle.bigint_groupby(keys)
    exp = oracle.grouped_sum_f64_exact(gids, vals, ng, scale_pow=43)
    assert np.array_equal(out[0]["values"], by_gid)
    assert np.array_equal(out[1]["values"], oracle.grouped_count(gids, ng))
    assert np.array_equal(out[2]["values"], exp)


def test_set_builder_sparse_fallback(sess, ops):
    """Key range beyond the bitmap bound falls back to the positional index
    with identical semi results."""
    r = np.random.default_rng(700)
    bk = r.choice(np.arange(0, 2**40, 2**20), 5000, replace=False).astype(np.int64)
    pk = np.concatenate([bk[:700], r.integers(0, 2**40, 9000)]).astype(np.int64)
    bridge = ops.JoinBridge(sess)
    b = ops.set_builder(sess, bridge, [ops.TG_BIGINT], 0)
    b.add_input(ops.page_from_numpy([bk]))
    b.drain()
    sj = ops.semi_join(sess, bridge, 0)
    sj.add_input(ops.page_from_numpy([pk]))
    out, _ = sj.get_output()
    b.close()
    sj.close()
    bridge.close()
    exp = np.isin(pk, bk)
    assert np.array_equal(out[-1]["values"].astype(bool), exp)
