"""Q6 (scan/filter/scalar-agg) through the operator pipeline, vs the oracle
composition and the public TPC-H Q6 answer @SF1 (revenue 123141078.2283).
Tolerance: count exact; revenue 1e-9 relative (scalar f64 atomic sum)."""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu


def q6_reference(sf, n_orders=None):
    li = oracle.gen_lineitem(sf, 1, n_orders)
    m = ((li["shipdate"] >= 8766) & (li["shipdate"] < 9131) &
         (li["discount"] >= 0.05) & (li["discount"] <= 0.07) &
         (li["quantity"] < 24))
    return float(np.sum(li["extendedprice"][m] * li["discount"][m])), int(m.sum())


def test_q6_sf001():
    import trino_amd
    from trino_amd import tpch_queries
    s = trino_amd.Session(0)
    try:
        got = tpch_queries.q6_gpu(s, 0.01)
        rev, cnt = q6_reference(0.01, 15000)
        assert got["rows"] == cnt
        assert abs(got["revenue"] - rev) <= 1e-9 * abs(rev)
    finally:
        s.close()


def test_q6_sf1_official_answer():
    import trino_amd
    from trino_amd import tpch_queries
    s = trino_amd.Session(0)
    try:
        got = tpch_queries.q6_gpu(s, 1.0)
        assert abs(got["revenue"] - 123141078.2283) < 0.01
    finally:
        s.close()
