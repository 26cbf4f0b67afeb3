"""Structural pins: SF1 generator columns vs the reference's own committed
statistics fixtures (plugin/trino-tpch/.../statistics/sf1.0/*.json —
rowCount / min / max / distinctValuesCount per column). These validate the
streams at full SF1 scale, beyond the per-row sf0.01 dataset pins."""
import json
import os

import numpy as np
import pytest

import oracle

HERE = os.path.dirname(__file__)


@pytest.fixture(scope="module")
def stats():
    with open(os.path.join(HERE, "golden", "tpch_stats.json")) as f:
        return json.load(f)["stats"]["sf1.0"]


def col(stats, table, name):
    return stats[table]["columns"][name]


def check(c, arr, distinct=True):
    assert int(c["min"]) == int(arr.min())
    assert int(c["max"]) == int(arr.max())
    if distinct and c.get("distinctValuesCount") is not None:
        assert int(c["distinctValuesCount"]) == len(np.unique(arr))


def test_part_stats(stats):
    p = oracle.gen_part2(1.0)
    assert stats["part"]["rowCount"] == len(p["partkey"])
    check(col(stats, "part", "p_size"), p["size"])
    # p_retailprice is DOUBLE in the fixture
    c = col(stats, "part", "p_retailprice")
    rp = p["retail_cents"] / 100.0
    assert abs(float(c["min"]) - rp.min()) < 1e-9
    assert abs(float(c["max"]) - rp.max()) < 1e-9
    assert int(c["distinctValuesCount"]) == len(np.unique(p["retail_cents"]))
    # name distinct count (199,997 at SF1: three collisions)
    names = [tuple(x) for x in p["name_ids"]]
    assert int(col(stats, "part", "p_name")["distinctValuesCount"]) == len(set(names))
    assert len(np.unique(p["brand"])) == 25
    assert len(np.unique(p["type_id"])) == 150
    assert len(np.unique(p["container"])) == 40


def test_partsupp_stats(stats):
    ps = oracle.gen_partsupp(1.0)
    assert stats["partsupp"]["rowCount"] == len(ps["partkey"])
    check(col(stats, "partsupp", "ps_availqty"), ps["availqty"])
    c = col(stats, "partsupp", "ps_supplycost")
    sc = ps["supplycost_cents"]
    assert abs(float(c["min"]) - sc.min() / 100.0) < 1e-9
    assert abs(float(c["max"]) - sc.max() / 100.0) < 1e-9
    assert int(c["distinctValuesCount"]) == len(np.unique(sc))


def test_supplier_stats(stats):
    s = oracle.gen_supplier2(1.0)
    assert stats["supplier"]["rowCount"] == len(s["suppkey"])
    check(col(stats, "supplier", "s_nationkey"), s["nationkey"])
    c = col(stats, "supplier", "s_acctbal")
    ab = s["acctbal_cents"]
    assert abs(float(c["min"]) - ab.min() / 100.0) < 1e-9
    assert abs(float(c["max"]) - ab.max() / 100.0) < 1e-9
    assert int(c["distinctValuesCount"]) == len(np.unique(ab))


def test_customer_stats(stats):
    cgen = oracle.gen_customer(1.0)
    assert stats["customer"]["rowCount"] == len(cgen["custkey"])
    check(col(stats, "customer", "c_nationkey"), cgen["nationkey"])
    c = col(stats, "customer", "c_acctbal")
    ab = cgen["acctbal_cents"]
    assert abs(float(c["min"]) - ab.min() / 100.0) < 1e-9
    assert abs(float(c["max"]) - ab.max() / 100.0) < 1e-9
    assert int(c["distinctValuesCount"]) == len(np.unique(ab))


def test_orders_stats(stats):
    o = oracle.gen_orders3(1.0)
    assert stats["orders"]["rowCount"] == len(o["orderkey"])
    check(col(stats, "orders", "o_orderkey"), o["orderkey"])
    check(col(stats, "orders", "o_custkey"), o["custkey"])
    check(col(stats, "orders", "o_orderdate"), o["orderdate"])
    c = col(stats, "orders", "o_totalprice")
    tp = o["totalprice_cents"]
    assert abs(float(c["min"]) - tp.min() / 100.0) < 1e-9
    assert abs(float(c["max"]) - tp.max() / 100.0) < 1e-9
    assert int(c["distinctValuesCount"]) == len(np.unique(tp))
    # clerk pool: 1000 distinct at SF1
    assert int(col(stats, "orders", "o_clerk")["distinctValuesCount"]) == \
        len(np.unique(o["clerk"]))


def test_lineitem_stats(stats):
    li = oracle.gen_lineitem(1.0, columns=["shipdate", "commitdate",
                                           "receiptdate", "partkey",
                                           "suppkey", "extendedprice"])
    assert stats["lineitem"]["rowCount"] == len(li["shipdate"])
    check(col(stats, "lineitem", "l_shipdate"), li["shipdate"])
    check(col(stats, "lineitem", "l_commitdate"), li["commitdate"])
    check(col(stats, "lineitem", "l_receiptdate"), li["receiptdate"])
    check(col(stats, "lineitem", "l_partkey"), li["partkey"])
    check(col(stats, "lineitem", "l_suppkey"), li["suppkey"])
    c = col(stats, "lineitem", "l_extendedprice")
    ep = li["extendedprice"]
    assert abs(float(c["min"]) - ep.min()) < 1e-9
    assert abs(float(c["max"]) - ep.max()) < 1e-9
