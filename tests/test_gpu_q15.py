"""TPC-H Q15 on device vs the reference's SF1 answer fixture: the top
supplier by 3-month revenue (suppkey + exact revenue; the fixture's
address/phone columns are unpinned text and not compared)."""
import json
import os

import pytest

pytestmark = pytest.mark.gpu


def test_q15_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["all_answers_sf1"]["rows"]["q15"]
    s = trino_amd.Session(0)
    try:
        r = q.q15_gpu(s, 1.0)
    finally:
        s.close()
    assert len(r["suppkey"]) == len(fx)
    for i, row in enumerate(fx):
        assert int(r["suppkey"][i]) == int(row[0])
        assert r["s_name"][i] == row[1]
        assert abs(float(r["total_revenue"][i]) - float(row[4])) < 5e-5
