"""Q4 (EXISTS semi join + grouped count) through the operator pipeline,
pinned to the public TPC-H Q4 answer @SF1. Counts bit-exact."""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

Q4_SF1 = [10594, 10476, 10410, 10556, 10487]  # 1-URGENT..5-LOW


def test_q4_sf1_official_answer():
    import trino_amd
    from trino_amd import tpch_queries
    s = trino_amd.Session(0)
    try:
        got = tpch_queries.q4_gpu(s, 1.0)
        assert got["priority"].tolist() == [0, 1, 2, 3, 4]
        assert got["count"].tolist() == Q4_SF1
    finally:
        s.close()


def test_q4_sf001_vs_oracle():
    import trino_amd
    from trino_amd import tpch_queries
    s = trino_amd.Session(0)
    try:
        got = tpch_queries.q4_gpu(s, 0.01)
    finally:
        s.close()
    o = oracle.gen_orders(0.01, 1, 15000)
    li = oracle.gen_lineitem(0.01, 1, 15000)
    late = np.unique(li["orderkey"][li["commitdate"] < li["receiptdate"]])
    sel = (o["orderdate"] >= 8582) & (o["orderdate"] < 8674) & np.isin(o["orderkey"], late)
    exp = np.bincount(o["orderpriority"][sel], minlength=5)
    got_c = np.zeros(5, np.int64)
    got_c[got["priority"]] = got["count"]
    assert np.array_equal(got_c, exp)
