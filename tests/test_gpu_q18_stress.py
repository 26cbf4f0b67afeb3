"""Q18-shaped high-cardinality pipeline stress (large-orders query,
reference plugin/trino-tpch Q18 shape): SF1 lineitem grouped by orderkey —
~1.5M groups — HAVING sum(l_quantity) > threshold, then join back against
lineitem and re-aggregate.

This is the stress test for the group table's growth path
(HashAggOp::grow_if_needed + k_gt_rehash at ~1.5M groups; the operator
starts at 2^16 group capacity) and for the device group-id remap
(run_argsort_i64) at a size where the device radix sort is engaged.

Parity: numpy composition of the same plan over the oracle-identical
device-generated columns (generator parity is pinned elsewhere:
tests/test_tpchgen_oracle.py).
"""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

QTY_THRESHOLD = 300.0  # Q18 parameter


def test_q18_shape_high_cardinality():
    import trino_amd
    from trino_amd import ops

    s = trino_amd.Session(0)
    try:
        li = s.tpch_lineitem(1.0, with_orderkey=True)
        n = li.row_count
        lpage = ops.page_from_device(s, ([(li.orderkey, ops.TG_BIGINT),
                                          (li.quantity, ops.TG_DOUBLE)], n))
        # stage 1: group by orderkey (≈1.5M groups), sum quantity
        agg1 = ops.hash_aggregation(s, [0], [ops.TG_BIGINT],
                                    [(ops.AGG_SUM_F64, 1)])
        agg1.add_input(lpage)
        agg1.finish()
        g1, _ = agg1.get_output()

        # host reference over the same device-generated inputs
        ok = np.empty(n, np.int64)
        qty = np.empty(n, np.float64)
        trino_amd.copy_dtoh(s, ok, li.orderkey)
        trino_amd.copy_dtoh(s, qty, li.quantity)
        keys, inv = np.unique(ok, return_inverse=True)
        sums = np.zeros(len(keys))
        np.add.at(sums, inv, qty)
        # group ids must follow first-occurrence row order
        first_pos = np.full(len(keys), n, np.int64)
        np.minimum.at(first_pos, inv, np.arange(n))
        order = np.argsort(first_pos, kind="stable")
        assert len(g1[0]["values"]) == len(keys)
        assert np.array_equal(g1[0]["values"], keys[order])
        np.testing.assert_allclose(g1[1]["values"], sums[order], rtol=1e-12)

        # stage 2: HAVING sum(quantity) > 300 (filter over the agg output)
        f = ops.expr(("col", 1), ("f64", QTY_THRESHOLD), "gt")
        fp = ops.filter_project(s, f, [ops.expr(("col", 0))], [ops.TG_BIGINT])
        fp.add_input(ops.page_from_numpy([np.asarray(g1[0]["values"]),
                                          np.asarray(g1[1]["values"])]))
        fp.finish()
        big, _ = fp.get_output()
        exp_big = keys[order][sums[order] > QTY_THRESHOLD]
        if big is None:
            assert len(exp_big) == 0
            return
        assert np.array_equal(big[0]["values"], exp_big)

        # stage 3: semi-join lineitem against the qualifying orderkeys,
        # re-aggregate the surviving rows
        bridge = ops.JoinBridge(s)
        b = ops.hash_builder(s, bridge, [ops.TG_BIGINT], [0], [])
        b.add_input(ops.page_from_numpy([big[0]["values"]]))
        b.drain()
        j = ops.lookup_join(s, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], [0, 1])
        j.add_input(ops.page_from_device(s, ([(li.orderkey, ops.TG_BIGINT),
                                              (li.quantity, ops.TG_DOUBLE)], n)))
        jp, _ = j.get_output()
        b.close()
        j.close()
        bridge.close()

        qual = np.isin(ok, exp_big)
        if jp is None:
            assert qual.sum() == 0
            return
        agg2 = ops.hash_aggregation(s, [0], [ops.TG_BIGINT],
                                    [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_F64, 1)])
        agg2.add_input(ops.page_from_numpy([np.asarray(jp[0]["values"]),
                                            np.asarray(jp[1]["values"])]))
        out = agg2.drain()[0]
        agg2.close()
        ok_q, qty_q = ok[qual], qty[qual]
        k2, inv2 = np.unique(ok_q, return_inverse=True)
        s2 = np.zeros(len(k2))
        np.add.at(s2, inv2, qty_q)
        c2 = np.bincount(inv2)
        fp2 = np.full(len(k2), n, np.int64)
        np.minimum.at(fp2, inv2, np.arange(len(ok_q)))
        # join emits probe rows in probe-row order, so first occurrence in
        # the joined stream == first occurrence among qualifying rows
        order2 = np.argsort(fp2, kind="stable")
        assert np.array_equal(out[0]["values"], k2[order2])
        assert np.array_equal(out[1]["values"], c2[order2])
        np.testing.assert_allclose(out[2]["values"], s2[order2], rtol=1e-12)
        s.tpch_lineitem_free(li)
    finally:
        s.close()
