"""Round-2 query pipelines (device, operator C-ABI) vs the reference's SF1
answer fixtures — same pins as the CPU oracle compositions in
test_oracle_queries.py, now through the GPU operator chains."""
import json
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(__file__)


@pytest.fixture(scope="module")
def fx():
    with open(os.path.join(HERE, "golden", "ref_fixtures.json")) as f:
        return json.load(f)["all_answers_sf1"]["rows"]


@pytest.fixture(scope="module")
def sess():
    import trino_amd
    s = trino_amd.Session(0)
    yield s
    s.close()


def test_q02(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q2_gpu(sess, 1.0)
    exp = fx["q02"]
    assert len(r["rows"]) == len(exp)
    for g, e in zip(r["rows"], exp):
        assert abs(g["acctbal_cents"] / 100.0 - float(e[0])) < 1e-9
        assert g["s_name"] == e[1] and g["n_name"] == e[2]
        assert g["partkey"] == int(e[3]) and g["mfgr"] == e[4]
        assert g["address"] == e[5] and g["phone"] == e[6]
        assert g["comment"] == e[7]


def test_q09(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q9_gpu(sess, 1.0)
    exp = [(x[0], int(x[1]), float(x[2])) for x in fx["q09"]]
    assert len(r["rows"]) == len(exp)
    for g, e in zip(r["rows"], exp):
        assert g[0] == e[0] and g[1] == e[1] and abs(g[2] - e[2]) < 0.002


def test_q10(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q10_gpu(sess, 1.0)
    exp = fx["q10"]
    assert len(r["rows"]) == len(exp)
    for g, e in zip(r["rows"], exp):
        assert g["custkey"] == int(e[0]) and g["name"] == e[1]
        assert abs(g["revenue"] - float(e[2])) < 0.002
        assert abs(g["acctbal_cents"] / 100.0 - float(e[3])) < 1e-9
        assert g["n_name"] == e[4] and g["address"] == e[5]
        assert g["phone"] == e[6] and g["comment"] == e[7]


def test_q11(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q11_gpu(sess, 1.0)
    exp = [(int(x[0]), float(x[1])) for x in fx["q11"]]
    assert len(r["partkey"]) == len(exp)
    for pk, cents, e in zip(r["partkey"], r["value_cents"], exp):
        assert int(pk) == e[0] and abs(cents / 100.0 - e[1]) < 0.005


def test_q13(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q13_gpu(sess, 1.0)
    got = list(zip(r["c_count"].tolist(), r["custdist"].tolist()))
    assert got == [(int(x[0]), int(x[1])) for x in fx["q13"]]


def test_q16(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q16_gpu(sess, 1.0)
    exp = [(x[0], x[1], int(x[2]), int(x[3])) for x in fx["q16"]]
    assert r["rows"] == exp


def test_q17(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q17_gpu(sess, 1.0)
    assert abs(r["avg_yearly"] - float(fx["q17"][0][0])) < 1e-4


def test_q19(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q19_gpu(sess, 1.0)
    assert abs(r["revenue"] - float(fx["q19"][0][0])) < 1e-3


def test_q20(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q20_gpu(sess, 1.0)
    exp = fx["q20"]
    assert len(r["names"]) == len(exp)
    for nm, ad, e in zip(r["names"], r["addresses"], exp):
        assert nm == e[0] and ad == e[1]


def test_q21(sess, fx):
    from trino_amd import tpch_queries as tq
    r = tq.q21_gpu(sess, 1.0)
    exp = [(x[0], int(x[1])) for x in fx["q21"]]
    got = list(zip(r["names"], r["numwait"].tolist()))
    assert got == exp
