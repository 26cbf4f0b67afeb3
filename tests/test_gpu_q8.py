"""TPC-H Q8 on device vs the reference's SF1 answer fixture."""
import json
import os

import pytest

pytestmark = pytest.mark.gpu


def test_q8_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["all_answers_sf1"]["rows"]["q08"]
    s = trino_amd.Session(0)
    try:
        r = q.q8_gpu(s, 1.0)
    finally:
        s.close()
    for i, row in enumerate(fx):
        assert int(r["o_year"][i]) == int(row[0])
        assert abs(float(r["mkt_share"][i]) - float(row[1])) < 5e-7
