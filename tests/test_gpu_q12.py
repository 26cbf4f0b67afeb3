"""TPC-H Q12 end-to-end on device: shipmode stream (seed pinned by the
reference's canonical rows + Q12 fixture — tests/test_tpchgen_oracle.py)
through filter -> join -> conditional-count aggregation. The SF1 output must
equal the reference's own answer fixture exactly (tests/golden/
ref_fixtures.json <- hive_tpch/q12.result)."""
import json
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_q12_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["q12_sf1"]["rows"]
    s = trino_amd.Session(0)
    try:
        r = q.q12_gpu(s, 1.0)
        assert r["shipmode"].tolist() == [4, 6]          # MAIL, SHIP
        assert r["high"].tolist() == [fx[0]["high"], fx[1]["high"]]
        assert r["low"].tolist() == [fx[0]["low"], fx[1]["low"]]
    finally:
        s.close()
