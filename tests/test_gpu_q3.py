"""Q3 end-to-end on device (customer⨝orders⨝lineitem + grouped revenue),
BASELINE config 3, vs an independent numpy+oracle composition and the public
TPC-H Q3 answer @SF1.

Tolerances: row counts / keys / dates bit-exact; revenue within 1e-6 relative
(device atomicAdd f64 group sums; each group sums <= ~7 products).
"""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

D = 9204  # 1995-03-15


def q3_reference(sf, n_orders=None, n_cust=None):
    cust = oracle.gen_customer(sf, 1, n_cust)
    orders = oracle.gen_orders(sf, 1, n_orders)
    li = oracle.gen_lineitem(sf, 1, n_orders)
    building = set(cust["custkey"][cust["mktsegment"] == 1].tolist())
    omask = (orders["orderdate"] < D) & np.isin(orders["custkey"],
                                                np.fromiter(building, np.int64))
    okeys = orders["orderkey"][omask]
    odates = dict(zip(okeys.tolist(), orders["orderdate"][omask].tolist()))
    lmask = (li["shipdate"] > D) & np.isin(li["orderkey"], okeys)
    rev = {}
    for ok, ep, di in zip(li["orderkey"][lmask].tolist(),
                          li["extendedprice"][lmask], li["discount"][lmask]):
        rev[ok] = rev.get(ok, 0.0) + ep * (1.0 - di)
    return {ok: (r, odates[ok]) for ok, r in rev.items()}


@pytest.fixture(scope="module")
def sess():
    import trino_amd
    s = trino_amd.Session(0)
    yield s
    s.close()


def test_q3_sf001_vs_reference(sess):
    from trino_amd import tpch_queries
    got = tpch_queries.q3_gpu(sess, 0.01)
    exp = q3_reference(0.01, 15000, 1500)
    assert len(got["orderkey"]) == len(exp)
    for ok, od, rv in zip(got["orderkey"].tolist(), got["orderdate"].tolist(),
                          got["revenue"]):
        er, ed = exp[ok]
        assert ed == od
        assert abs(rv - er) <= 1e-6 * max(1.0, abs(er)), ok


Q3_SF1_ANSWER = [
    (2456423, 406181.0111, "1995-03-05"),
    (3459808, 405838.6989, "1995-03-04"),
    (492164, 390324.0610, "1995-02-19"),
    (1188320, 384537.9359, "1995-03-09"),
    (2435712, 378673.0558, "1995-02-26"),
    (4878020, 378376.7952, "1995-03-12"),
    (5521732, 375153.9215, "1995-03-13"),
    (2628192, 373133.3094, "1995-02-22"),
    (993600, 371407.4595, "1995-03-05"),
    (2300070, 367371.1452, "1995-03-13"),
]


def _epoch(iso):
    import datetime
    return (datetime.date.fromisoformat(iso) - datetime.date(1970, 1, 1)).days


def test_q3_sf1_official_answer(sess):
    from trino_amd import tpch_queries
    got = tpch_queries.q3_gpu(sess, 1.0)
    assert len(got["top10"]) == 10
    for (gok, grev, god, gpri), (eok, erev, edate) in zip(got["top10"], Q3_SF1_ANSWER):
        assert gok == eok
        assert god == _epoch(edate)
        assert gpri == 0
        assert abs(grev - erev) < 0.01
