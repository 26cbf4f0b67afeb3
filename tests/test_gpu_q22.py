"""TPC-H Q22 end-to-end on device vs the reference's SF1 answer fixture
(hive_tpch/q22.result): country codes from the pinned customer nationkey
stream, balances exact in cents, NOT EXISTS via the set-builder bitmap."""
import json
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_q22_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["all_answers_sf1"]["rows"]["q22"]
    s = trino_amd.Session(0)
    try:
        r = q.q22_gpu(s, 1.0)
    finally:
        s.close()
    assert len(r["cntrycode"]) == len(fx)
    for i, row in enumerate(fx):
        assert int(r["cntrycode"][i]) == int(row[0])
        assert int(r["numcust"][i]) == int(row[1])
        assert int(r["totacctbal_cents"][i]) == round(float(row[2]) * 100)
