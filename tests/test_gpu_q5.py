"""TPC-H Q5 on device vs the reference's SF1 answer fixture: ASIA-region
revenue with customer/supplier nation equality, over the pinned customer and
supplier nationkey streams and the partsupp-bridge l_suppkey."""
import json
import os

import pytest

pytestmark = pytest.mark.gpu


def test_q5_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["all_answers_sf1"]["rows"]["q05"]
    s = trino_amd.Session(0)
    try:
        r = q.q5_gpu(s, 1.0)
    finally:
        s.close()
    assert len(r["n_name"]) == len(fx)
    for i, row in enumerate(fx):
        assert r["n_name"][i] == row[0]
        assert abs(float(r["revenue"][i]) - float(row[1])) < 5e-5
