"""Q1 through the GENERIC operator pipeline (filter_project -> hash
aggregation with grouped AVG and exact fixed-point SUMs) cross-validated
against the fused Q1 kernel and the oracle.

The fused kernel and the generic projection compute the same IEEE
expressions (ep*(1-d), that*(1+tax); -ffp-contract=off matches Java), and
both exact-sum paths accumulate the same 2^43/2^59-scaled integers — so the
SUM columns must be BIT-EQUAL across the two engines. AVG goes through the
generic (count,sum) state and compares at 1e-12.
"""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

CUTOFF = 10471  # shipdate <= 1998-09-02


def test_generic_pipeline_matches_fused():
    import trino_amd
    from trino_amd import ops

    s = trino_amd.Session(0)
    try:
        li = s.tpch_lineitem(0.2)
        n = li.row_count
        fused = s.q1(li, CUTOFF)

        lpage = ops.page_from_device(s, ([(li.shipdate, ops.TG_INTEGER),
                                          (li.quantity, ops.TG_DOUBLE),
                                          (li.extendedprice, ops.TG_DOUBLE),
                                          (li.discount, ops.TG_DOUBLE),
                                          (li.tax, ops.TG_DOUBLE),
                                          (li.returnflag, ops.TG_TINYINT),
                                          (li.linestatus, ops.TG_TINYINT)], n))
        f = ops.expr(("col", 0), ("i64", CUTOFF), "le")
        projs = [ops.expr(("col", 5)), ops.expr(("col", 6)),
                 ops.expr(("col", 1)), ops.expr(("col", 2)),
                 ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul"),
                 ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul",
                          ("f64", 1.0), ("col", 4), "add", "mul"),
                 ops.expr(("col", 3))]
        fp = ops.filter_project(s, f, projs,
                                [ops.TG_TINYINT, ops.TG_TINYINT, ops.TG_DOUBLE,
                                 ops.TG_DOUBLE, ops.TG_DOUBLE, ops.TG_DOUBLE,
                                 ops.TG_DOUBLE])
        fp.add_input(lpage)
        fp.finish()
        from trino_amd.tpch_queries import _take_device_page
        sel = _take_device_page(s, fp)

        # sum_qty is integer-valued (scale 0 suffices up to 2^52); prices on
        # the 2^-43 grid (>= 2^9); discount on 2^-59
        agg = ops.hash_aggregation(s, [0, 1], [ops.TG_TINYINT, ops.TG_TINYINT],
                                   [(ops.AGG_SUM_F64_EXACT, 2, 0),
                                    (ops.AGG_SUM_F64_EXACT, 3, 43),
                                    (ops.AGG_SUM_F64_EXACT, 4, 43),
                                    (ops.AGG_SUM_F64_EXACT, 5, 43),
                                    (ops.AGG_AVG_F64, 2),
                                    (ops.AGG_AVG_F64, 3),
                                    (ops.AGG_AVG_F64, 6),
                                    (ops.AGG_COUNT_STAR, -1)])
        agg.add_input(sel)
        out = agg.drain()[0]
        fp.close()
        agg.close()

        rf = np.asarray(out[0]["values"])
        ls = np.asarray(out[1]["values"])
        # fused result combos: (rf_id * 2 + ls_id), rf A/N/R=0/1/2, ls F/O=0/1
        order = np.argsort(rf * 2 + ls)
        for j, (name, fused_arr) in enumerate((
                ("sum_qty", fused.sum_qty), ("sum_base", fused.sum_base),
                ("sum_disc_price", fused.sum_disc_price),
                ("sum_charge", fused.sum_charge))):
            got = np.asarray(out[2 + j]["values"])[order]
            combos = (rf * 2 + ls)[order]
            exp = np.array([fused_arr[c] for c in combos])
            assert np.array_equal(got, exp), name      # bit-equal exact sums
        avg_qty = np.asarray(out[6]["values"])[order]
        avg_price = np.asarray(out[7]["values"])[order]
        avg_disc = np.asarray(out[8]["values"])[order]
        cnt = np.asarray(out[9]["values"])[order]
        combos = (rf * 2 + ls)[order]
        np.testing.assert_allclose(avg_qty, [fused.avg_qty[c] for c in combos], rtol=1e-12)
        np.testing.assert_allclose(avg_price, [fused.avg_price[c] for c in combos], rtol=1e-12)
        # AVG state is a plain atomicAdd f64 sum; ~3e5 addends of ~0.05 give
        # ~1.5e-12 relative wobble vs the fused kernel's exact sum
        np.testing.assert_allclose(avg_disc, [fused.avg_disc[c] for c in combos], rtol=1e-11)
        assert np.array_equal(cnt, [fused.count[c] for c in combos])
        s.tpch_lineitem_free(li)
    finally:
        s.close()
