"""TPC-H Q18 end-to-end on device vs the reference's own SF1 answer fixture
(tests/golden/ref_fixtures.json <- hive_tpch/q18.result): c_name, keys,
dates, o_totalprice to the cent (generator tp_cents = dbgen mk_order integer
truncation), sum(l_quantity)."""
import datetime
import json
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_q18_sf1_exact():
    import trino_amd
    from trino_amd import tpch_queries as q

    fx = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "ref_fixtures.json")))["q18_sf1"]["rows"]
    s = trino_amd.Session(0)
    try:
        r = q.q18_gpu(s, 1.0)
    finally:
        s.close()
    assert len(r["orderkey"]) == len(fx)
    epoch = datetime.date(1970, 1, 1)
    for i, e in enumerate(fx):
        assert r["c_name"][i] == e["c_name"]
        assert int(r["custkey"][i]) == e["custkey"]
        assert int(r["orderkey"][i]) == e["orderkey"]
        assert (epoch + datetime.timedelta(days=int(r["orderdate"][i]))
                ).isoformat() == e["orderdate"]
        assert int(r["totalprice_cents"][i]) == round(float(e["totalprice"]) * 100)
        assert int(r["sum_qty"][i]) == int(float(e["sum_qty"]))
