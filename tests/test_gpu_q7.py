"""TPC-H Q7 on device vs the reference's SF1 answer fixture."""
import json
import os

import pytest

pytestmark = pytest.mark.gpu


def test_q7_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["all_answers_sf1"]["rows"]["q07"]
    s = trino_amd.Session(0)
    try:
        r = q.q7_gpu(s, 1.0)
    finally:
        s.close()
    assert len(r["l_year"]) == len(fx)
    for i, row in enumerate(fx):
        assert r["supp_nation"][i] == row[0]
        assert r["cust_nation"][i] == row[1]
        assert int(r["l_year"][i]) == int(row[2])
        assert abs(float(r["revenue"][i]) - float(row[3])) < 5e-5
