"""trino_amd.tpch_queries — host-side drivers for the covered TPC-H plans,
composed from the C-ABI operators exactly as LocalExecutionPlanner chains the
reference's operator factories. Product code: device data
stays in HBM across the pipeline (pages flow between operators on_device=1);
only final grouped results come to the host.

Q3 (testing/trino-benchmark-queries/.../tpch/q03.sql):
  SELECT l_orderkey, sum(l_extendedprice*(1-l_discount)) AS revenue,
         o_orderdate, o_shippriority
  FROM customer, orders, lineitem
  WHERE c_mktsegment='BUILDING' AND c_custkey=o_custkey
    AND l_orderkey=o_orderkey AND o_orderdate < DATE '1995-03-15'
    AND l_shipdate > DATE '1995-03-15'
  GROUP BY l_orderkey, o_orderdate, o_shippriority
  ORDER BY revenue DESC, o_orderdate LIMIT 10
Plan shape (reference: broadcast build customer -> join orders -> build ->
probe lineitem -> hash agg -> TopN). TopN is §8(f) next-row: the final
ORDER BY ... LIMIT 10 here runs on the grouped output (tiny) on the host.
"""
import ctypes
import os
import time

import numpy as np

from . import _lib, LineitemCols, Session, copy_dtoh
from . import ops

DATE_1995_03_15 = 9204


def q3_prepare(session, sf, order_start=1, order_count=None, cust_start=1, cust_count=None,
               lineitem_order_start=None, lineitem_order_count=None):
    """Generate Q3's device-resident inputs (NOT part of the timed query)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    if cust_count is None:
        cust_count = int(150_000 * sf)
    if lineitem_order_start is None:
        lineitem_order_start = order_start
        lineitem_order_count = order_count
    cust_ck = _device_buffer(session, cust_count * 8)
    cust_ms = _device_buffer(session, cust_count)
    _check_lib(_lib.tg_tpch_gen_customer(session._h, sf,
                                         cust_start, cust_count, cust_ck, cust_ms,
                                         None, None))
    o_ok = _device_buffer(session, order_count * 8)
    o_ck = _device_buffer(session, order_count * 8)
    o_od = _device_buffer(session, order_count * 4)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf,
                                       order_start, order_count, o_ok, o_ck, o_od, None))
    li = session.tpch_lineitem(sf, lineitem_order_start, lineitem_order_count,
                               with_orderkey=True)
    return dict(cust_ck=cust_ck, cust_ms=cust_ms, cust_count=cust_count,
                o_ok=o_ok, o_ck=o_ck, o_od=o_od, order_count=order_count, li=li)


def q3_release(session, inp):
    session.tpch_lineitem_free(inp["li"])
    for k in ("cust_ck", "cust_ms", "o_ok", "o_ck", "o_od"):
        _device_free(session, inp[k])


def q3_execute(session, inp, download_groups=True, cust_key_exchange=None):
    """The timed Q3 pipeline over prepared device inputs.

    cust_key_exchange: distributed hook (trino_amd.dist.gather_union bound to
    the process group) — receives this rank's filtered BUILDING custkeys and
    returns the union across ranks (broadcast/replicated join build,
    DetermineJoinDistributionType BROADCAST). None = single-GPU."""
    cust_ck, cust_ms, cust_count = inp["cust_ck"], inp["cust_ms"], inp["cust_count"]
    o_ok, o_ck, o_od, order_count = inp["o_ok"], inp["o_ck"], inp["o_od"], inp["order_count"]
    li = inp["li"]

    t0 = time.time()
    # ---- stage 1: customer filter (mktsegment = 'BUILDING' = id 1) ----
    cpage = ops.page_from_device(session, ([(cust_ck.value, ops.TG_BIGINT),
                                            (cust_ms.value, ops.TG_TINYINT)], cust_count))
    f1 = ops.filter_project(session, ops.expr(("col", 1), ("i64", 1), "eq"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    f1.add_input(cpage)
    f1.finish()
    bridge1 = ops.JoinBridge(session)
    b1 = ops.hash_builder(session, bridge1, [ops.TG_BIGINT], [0], [])
    keep_exchange = None
    if cust_key_exchange is None:
        cust_sel = _take_device_page(session, f1)
        b1.add_input(cust_sel)
    elif getattr(cust_key_exchange, "device_resident", False):
        # BROADCAST build stays in HBM: tg buffer -> torch cuda tensor ->
        # RCCL all-gather over xGMI -> build input page (no host round-trip)
        cust_sel = _take_device_page(session, f1)
        t = cust_key_exchange(session, cust_sel)
        keep_exchange = t
        b1.add_input(ops.page_from_device(
            session, ([(t.data_ptr(), ops.TG_BIGINT)], t.numel())))
    else:
        host_sel, _ = f1.get_output()     # downloads the selected custkeys
        local_keys = host_sel[0]["values"] if host_sel else np.empty(0, np.int64)
        all_keys = cust_key_exchange(local_keys)
        b1.add_input(ops.page_from_numpy([np.ascontiguousarray(all_keys)]))
    b1.drain()

    # ---- stage 2: orders filter + join customers ----
    opage = ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                            (o_ck.value, ops.TG_BIGINT),
                                            (o_od.value, ops.TG_INTEGER)], order_count))
    f2 = ops.filter_project(session, ops.expr(("col", 2), ("i64", DATE_1995_03_15), "lt"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1)), ops.expr(("col", 2))],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_INTEGER])
    f2.add_input(opage)
    f2.finish()
    orders_sel = _take_device_page(session, f2)

    j1 = ops.lookup_join(session, bridge1, [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_INTEGER],
                         [1], [0, 2])      # probe key custkey; emit orderkey, orderdate
    j1.add_input(orders_sel)
    j1.finish()
    orders_building = _take_device_page(session, j1)

    bridge2 = ops.JoinBridge(session)
    ops.request_bitmap(bridge2)   # dynamic filter -> pushed into the scan below
    b2 = ops.hash_builder(session, bridge2, [ops.TG_BIGINT, ops.TG_INTEGER],
                          [0], [1])        # key orderkey; build output orderdate
    b2.add_input(orders_building)
    b2.drain()

    # ---- stage 3: lineitem filter+project, probe, aggregate ----
    lpage = ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                            (li.shipdate, ops.TG_INTEGER),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE)],
                                           li.row_count))
    # scan filter + FUSED dynamic filter (orderkey ∈ build keys): cuts the
    # probe input to BUILDING-customer orders before the join
    f3 = ops.filter_project_df(session,
                               ops.expr(("col", 1), ("i64", DATE_1995_03_15), "gt"),
                               [ops.expr(("col", 0)),
                                ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul")],
                               [ops.TG_BIGINT, ops.TG_DOUBLE], bridge2, 0)
    f3.add_input(lpage)
    f3.finish()
    li_sel = _take_device_page(session, f3)

    j2 = ops.lookup_join(session, bridge2, [ops.TG_BIGINT, ops.TG_DOUBLE],
                         [0], [0, 1])      # emit orderkey, discprice (+ build orderdate)
    probe_ms = _timed(session, lambda: j2.add_input(li_sel))
    j2.finish()
    joined = _take_device_page(session, j2)
    probe_rows = li_sel.position_count
    match_rows = joined.position_count

    # revenue = sum(ep*(1-disc)): products >= ~810 (= 2^9 < v < 2^11) sit on
    # the 2^-43 grid -> TG_AGG_SUM_F64_EXACT is exact AND order-independent
    # (DESIGN.md §4's Q1 argument applied to the generic operator)
    agg = ops.hash_aggregation(session, [0, 2], [ops.TG_BIGINT, ops.TG_INTEGER],
                               [(ops.AGG_SUM_F64_EXACT, 1, 43)])
    agg.add_input(joined)
    agg.finish()
    agg_page = _take_device_page(session, agg)

    # TopN: ORDER BY revenue DESC, o_orderdate ASC LIMIT 10 (TopNOperator)
    top = ops.topn(session, [ops.TG_BIGINT, ops.TG_INTEGER, ops.TG_DOUBLE],
                   [2, 1], [1, 0], 10)
    top.add_input(agg_page)
    top_pages = top.drain()
    elapsed = time.time() - t0
    pages_full = ops._download_page(session, agg_page) if download_groups else None

    # cleanup (inputs stay alive for repeated execution)
    for op in (f1, b1, f2, j1, b2, f3, j2, agg, top):
        op.close()
    bridge1.close()
    bridge2.close()

    top10 = []
    if top_pages:
        tp = top_pages[0]
        top10 = [(int(tp[0]["values"][i]), float(tp[2]["values"][i]),
                  int(tp[1]["values"][i]), 0) for i in range(len(tp[0]["values"]))]
    out = pages_full
    if out is None:
        return dict(orderkey=None, orderdate=None, revenue=None,
                    top10=top10, elapsed=elapsed, probe_ms=probe_ms,
                    probe_rows=probe_rows, match_rows=match_rows)
    orderkey = out[0]["values"]
    orderdate = out[1]["values"]
    revenue = out[2]["values"]
    return dict(orderkey=orderkey, orderdate=orderdate, revenue=revenue,
                top10=top10, elapsed=elapsed, probe_ms=probe_ms,
                probe_rows=probe_rows, match_rows=match_rows)


# ---- small device-buffer helpers over the C ABI ----
_lib.tg_timer_start.restype = ctypes.c_int
_lib.tg_timer_start.argtypes = [ctypes.c_void_p]
_lib.tg_timer_stop.restype = ctypes.c_int
_lib.tg_timer_stop.argtypes = [ctypes.c_void_p, ctypes.c_void_p]


def _timed(session, fn):
    ms = ctypes.c_double()
    _check_lib(_lib.tg_timer_start(session._h))
    fn()
    _check_lib(_lib.tg_timer_stop(session._h, ctypes.byref(ms)))
    return ms.value


_lib.tg_device_malloc.restype = ctypes.c_int
_lib.tg_device_malloc.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64]
_lib.tg_device_free.restype = ctypes.c_int
_lib.tg_device_free.argtypes = [ctypes.c_void_p, ctypes.c_void_p]


def _check_lib(status):
    from . import _check
    _check(status)


def _device_buffer(session, nbytes):
    p = ctypes.c_void_p()
    _check_lib(_lib.tg_device_malloc(session._h, ctypes.byref(p), ctypes.c_int64(nbytes)))
    return p


def _device_free(session, p):
    _check_lib(_lib.tg_device_free(session._h, p))


def _take_device_page(session, op):
    """Fetch the operator's staged output page WITHOUT downloading: returns a
    device tg_page usable as the next operator's input (pointers remain owned
    by the producing operator; keep it open until the consumer is done)."""
    out = ops.TgPage()
    fin = ctypes.c_int(0)
    _check_lib(_lib.tg_operator_get_output(op._h, ctypes.byref(out), ctypes.byref(fin)))
    return out


def q3_gpu(session, sf, **kw):
    """prepare + execute + release (tests / one-shot use)."""
    inp = q3_prepare(session, sf, **kw)
    try:
        return q3_execute(session, inp)
    finally:
        q3_release(session, inp)


DATE_1994_01_01 = 8766
DATE_1995_01_01 = 9131


def q6_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q6 (testing/trino-benchmark-queries/.../tpch/q06.sql):
    SELECT sum(l_extendedprice*l_discount) FROM lineitem
    WHERE l_shipdate >= '1994-01-01' AND l_shipdate < '1995-01-01'
      AND l_discount BETWEEN .06-0.01 AND .06+0.01 AND l_quantity < 24
    Pipeline: filter+project -> scalar aggregation (AggregationOperator)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    li = session.tpch_lineitem(sf, order_start, order_count)
    t0 = time.time()
    lpage = ops.page_from_device(session, ([(li.shipdate, ops.TG_INTEGER),
                                            (li.quantity, ops.TG_DOUBLE),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE)],
                                           li.row_count))
    filt = ops.expr(("col", 0), ("i64", DATE_1994_01_01), "ge",
                    ("col", 0), ("i64", DATE_1995_01_01), "lt", "and",
                    ("col", 3), ("f64", 0.05), ("f64", 0.07), "between", "and",
                    ("col", 1), ("i64", 24), "lt", "and")
    fp = ops.filter_project(session, filt,
                            [ops.expr(("col", 2), ("col", 3), "mul")], [ops.TG_DOUBLE])
    fp.add_input(lpage)
    fp.finish()
    sel = _take_device_page(session, fp)
    # ep*disc >= ~45 (2^5): exact on the 2^-47 grid
    agg = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_F64_EXACT, 0, 47),
                                                (ops.AGG_COUNT_STAR, -1)])
    agg.add_input(sel)
    pages = agg.drain()
    elapsed = time.time() - t0
    fp.close()
    agg.close()
    session.tpch_lineitem_free(li)
    out = pages[0]
    return dict(revenue=float(out[0]["values"][0]), rows=int(out[1]["values"][0]),
                elapsed=elapsed)


DATE_1993_07_01 = 8582
DATE_1993_10_01 = 8674


def q4_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q4: SELECT o_orderpriority, count(*) FROM orders
    WHERE o_orderdate in [1993-07-01, 1993-10-01) AND EXISTS (lineitem with
    l_commitdate < l_receiptdate) GROUP BY o_orderpriority ORDER BY it.
    Pipeline: lineitem filter -> semi-join source build; orders filter ->
    semi join -> filter matched -> group-by count."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    o_ok = _device_buffer(session, order_count * 8)
    o_od = _device_buffer(session, order_count * 4)
    o_pri = _device_buffer(session, order_count)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, order_start, order_count,
                                       o_ok, None, o_od, o_pri))
    li = session.tpch_lineitem(sf, order_start, order_count,
                               with_orderkey=True, with_dates=True)
    t0 = time.time()
    # build side: lineitem orderkeys where commitdate < receiptdate
    lpage = ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                            (li.commitdate, ops.TG_INTEGER),
                                            (li.receiptdate, ops.TG_INTEGER)],
                                           li.row_count))
    f1 = ops.filter_project(session, ops.expr(("col", 1), ("col", 2), "lt"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    f1.add_input(lpage)
    f1.finish()
    late = _take_device_page(session, f1)
    bridge = ops.JoinBridge(session)
    # SetBuilderOperator: semi-join source needs membership only — the dense
    # orderkey range builds a bitmap instead of a 380M-row positional index
    b = ops.set_builder(session, bridge, [ops.TG_BIGINT], 0)
    b.add_input(late)
    b.drain()
    # probe side: orders in the date window
    opage = ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                            (o_od.value, ops.TG_INTEGER),
                                            (o_pri.value, ops.TG_TINYINT)],
                                           order_count))
    f2 = ops.filter_project(session,
                            ops.expr(("col", 1), ("i64", DATE_1993_07_01), "ge",
                                     ("col", 1), ("i64", DATE_1993_10_01), "lt", "and"),
                            [ops.expr(("col", 0)), ops.expr(("col", 2))],
                            [ops.TG_BIGINT, ops.TG_TINYINT])
    f2.add_input(opage)
    f2.finish()
    owin = _take_device_page(session, f2)
    sj = ops.semi_join(session, bridge, 0)
    sj.add_input(owin)
    sj.finish()
    marked = _take_device_page(session, sj)        # (orderkey, priority, matched)
    f3 = ops.filter_project(session, ops.expr(("col", 2), ("i64", 1), "eq"),
                            [ops.expr(("col", 1))], [ops.TG_TINYINT])
    f3.add_input(marked)
    f3.finish()
    exists = _take_device_page(session, f3)
    agg = ops.hash_aggregation(session, [0], [ops.TG_TINYINT],
                               [(ops.AGG_COUNT_STAR, -1)])
    agg.add_input(exists)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (f1, b, f2, sj, f3, agg):
        op.close()
    bridge.close()
    session.tpch_lineitem_free(li)
    for p in (o_ok, o_od, o_pri):
        _device_free(session, p)
    out = pages[0]
    pri = out[0]["values"]
    cnt = out[1]["values"]
    order = np.argsort(pri)
    return dict(priority=pri[order], count=cnt[order], elapsed=elapsed)


DATE_1995_09_01 = 9374
DATE_1995_10_01 = 9404


def q14_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q14: promo revenue percent over the Sep-1995 shipdate window.
    Pipeline: part build (key partkey -> p_type) ; lineitem filter+project ->
    join -> scalar sums (promo CASE via a type-range filter, and total)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    n_parts = int(200_000 * sf)
    p_pk = _device_buffer(session, n_parts * 8)
    p_ty = _device_buffer(session, n_parts * 2)   # SMALLINT: ids 0..149
    _check_lib(_lib.tg_tpch_gen_part(session._h, sf, 1, n_parts, p_pk, p_ty))
    li = session.tpch_lineitem(sf, order_start, order_count, with_partkey=True)
    t0 = time.time()
    ppage = ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                            (p_ty.value, ops.TG_SMALLINT)], n_parts))
    bridge = ops.JoinBridge(session)
    b = ops.hash_builder(session, bridge, [ops.TG_BIGINT, ops.TG_SMALLINT], [0], [1])
    b.add_input(ppage)
    b.drain()
    lpage = ops.page_from_device(session, ([(li.partkey, ops.TG_BIGINT),
                                            (li.shipdate, ops.TG_INTEGER),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE)],
                                           li.row_count))
    f = ops.filter_project(session,
                           ops.expr(("col", 1), ("i64", DATE_1995_09_01), "ge",
                                    ("col", 1), ("i64", DATE_1995_10_01), "lt", "and"),
                           [ops.expr(("col", 0)),
                            ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul")],
                           [ops.TG_BIGINT, ops.TG_DOUBLE])
    f.add_input(lpage)
    f.finish()
    sel = _take_device_page(session, f)
    j = ops.lookup_join(session, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], [1])
    j.add_input(sel)
    j.finish()
    joined = _take_device_page(session, j)     # (discprice, p_type)
    # total revenue
    a1 = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_F64_EXACT, 0, 43)])
    a1.add_input(joined)
    total = a1.drain()[0][0]["values"][0]
    # promo (CASE WHEN p_type LIKE 'PROMO%'): type ids 125..149
    f2 = ops.filter_project(session, ops.expr(("col", 1), ("i64", 125), "ge"),
                            [ops.expr(("col", 0))], [ops.TG_DOUBLE])
    f2.add_input(joined)
    f2.finish()
    promo_page = _take_device_page(session, f2)
    a2 = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_F64_EXACT, 0, 43)])
    a2.add_input(promo_page)
    promo = a2.drain()[0][0]["values"][0]
    elapsed = time.time() - t0
    for op in (b, f, j, a1, f2, a2):
        op.close()
    bridge.close()
    session.tpch_lineitem_free(li)
    for p in (p_pk, p_ty):
        _device_free(session, p)
    return dict(promo_revenue=100.0 * promo / total, elapsed=elapsed)


MODE_MAIL, MODE_SHIP = 4, 6   # dists.dss order REG AIR,AIR,RAIL,TRUCK,MAIL,FOB,SHIP


def q12_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q12 (testing/trino-benchmark-queries/.../tpch/q12.sql):
    per shipmode in (MAIL, SHIP), count late lines split by order priority
    (high = 1-URGENT/2-HIGH). Pins the L_SMODE stream: the SF1 answer must
    equal the reference fixture testing/trino-product-tests/.../hive_tpch/
    q12.result (MAIL 6202|9324, SHIP 6200|9262).
    Pipeline: orders build (orderkey -> priority); lineitem filter
    (mode IN + commit<receipt + ship<commit + receipt in 1994) -> join ->
    project high-flag -> group by shipmode."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    o_ok = _device_buffer(session, order_count * 8)
    o_pri = _device_buffer(session, order_count)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, order_start, order_count,
                                       o_ok, None, None, o_pri))
    li = session.tpch_lineitem(sf, order_start, order_count,
                               with_orderkey=True, with_dates=True,
                               with_shipmode=True)
    t0 = time.time()
    lpage = ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                            (li.shipmode, ops.TG_TINYINT),
                                            (li.shipdate, ops.TG_INTEGER),
                                            (li.commitdate, ops.TG_INTEGER),
                                            (li.receiptdate, ops.TG_INTEGER)],
                                           li.row_count))
    f = ops.filter_project(session,
                           ops.expr(("col", 1), ("i64", MODE_MAIL), "eq",
                                    ("col", 1), ("i64", MODE_SHIP), "eq", "or",
                                    ("col", 3), ("col", 4), "lt", "and",
                                    ("col", 2), ("col", 3), "lt", "and",
                                    ("col", 4), ("i64", DATE_1994_01_01), "ge", "and",
                                    ("col", 4), ("i64", DATE_1994_01_01 + 365), "lt", "and"),
                           [ops.expr(("col", 0)), ops.expr(("col", 1))],
                           [ops.TG_BIGINT, ops.TG_TINYINT])
    f.add_input(lpage)
    f.finish()
    sel = _take_device_page(session, f)
    # build the SMALL side (filtered late lines, ~0.03% of lineitem) and
    # probe orders — the join distribution the reference's optimizer picks
    bridge = ops.JoinBridge(session)
    b = ops.hash_builder(session, bridge, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b.add_input(sel)
    b.drain()
    j = ops.lookup_join(session, bridge, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    j.add_input(ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                                (o_pri.value, ops.TG_TINYINT)],
                                               order_count)))
    j.finish()
    joined = _take_device_page(session, j)     # (priority, shipmode)
    fp = ops.filter_project(session, None,
                            [ops.expr(("col", 1)),
                             ops.expr(("col", 0), ("i64", 1), "le")],
                            [ops.TG_TINYINT, ops.TG_DOUBLE])
    fp.add_input(joined)
    fp.finish()
    flagged = _take_device_page(session, fp)   # (shipmode, high 0/1)
    agg = ops.hash_aggregation(session, [0], [ops.TG_TINYINT],
                               [(ops.AGG_SUM_F64, 1), (ops.AGG_COUNT_STAR, -1)])
    agg.add_input(flagged)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (b, f, j, fp, agg):
        op.close()
    bridge.close()
    session.tpch_lineitem_free(li)
    for p in (o_ok, o_pri):
        _device_free(session, p)
    out = pages[0]
    mode = np.asarray(out[0]["values"])
    high = np.asarray(out[1]["values"]).astype(np.int64)
    cnt = np.asarray(out[2]["values"])
    order = np.argsort(mode)                    # MAIL(4) before SHIP(6)
    return dict(shipmode=mode[order], high=high[order],
                low=(cnt - high)[order], elapsed=elapsed)


def q18_gpu(session, sf, order_start=1, order_count=None, limit=100):
    """TPC-H Q18 (large-orders query): customers with orders whose total
    quantity exceeds 300. The SF1 answer must equal the reference fixture
    (hive_tpch/q18.result; o_totalprice to the cent via the generator's
    tp_cents column — dbgen mk_order truncation, verified on the canonical
    orders). Pipeline: lineitem group-by orderkey (sum qty ~1.5M groups, sum
    tp_cents) -> HAVING filter -> build -> probe orders -> TopN
    (totalprice DESC, orderdate ASC) -> c_name formatting host-side."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    o_ok = _device_buffer(session, order_count * 8)
    o_ck = _device_buffer(session, order_count * 8)
    o_od = _device_buffer(session, order_count * 4)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, order_start, order_count,
                                       o_ok, o_ck, o_od, None))
    li = session.tpch_lineitem(sf, order_start, order_count,
                               with_orderkey=True, with_totalprice=True)
    t0 = time.time()
    lpage = ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                            (li.quantity, ops.TG_DOUBLE),
                                            (li.tp_cents, ops.TG_BIGINT)],
                                           li.row_count))
    # lineitem is clustered by orderkey -> streaming aggregation (the
    # planner's StreamingAggregationOperator choice for grouped input); the
    # hash path costs ~700 ms at SF100's 150M groups, this is 3 linear passes
    agg1 = ops.streaming_aggregation(session, 0,
                                     [(ops.AGG_SUM_F64_EXACT, 1, 0),
                                      (ops.AGG_SUM_I64, 2)])
    agg1.add_input(lpage)
    agg1.finish()
    groups = _take_device_page(session, agg1)   # (orderkey, sumqty, totcents)
    f = ops.filter_project(session, ops.expr(("col", 1), ("f64", 300.0), "gt"),
                           [ops.expr(("col", 0)), ops.expr(("col", 1)),
                            ops.expr(("col", 2))],
                           [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_BIGINT])
    f.add_input(groups)
    f.finish()
    big = _take_device_page(session, f)
    bridge = ops.JoinBridge(session)
    b = ops.hash_builder(session, bridge,
                         [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_BIGINT], [0], [1, 2])
    b.add_input(big)
    b.drain()
    j = ops.lookup_join(session, bridge,
                        [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_INTEGER],
                        [0], [0, 1, 2])
    j.add_input(ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                                (o_ck.value, ops.TG_BIGINT),
                                                (o_od.value, ops.TG_INTEGER)],
                                               order_count)))
    j.finish()
    matched = _take_device_page(session, j)  # (okey, ckey, odate, sumqty, totcents)
    top = ops.topn(session, [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_INTEGER,
                             ops.TG_DOUBLE, ops.TG_BIGINT],
                   [4, 2], [1, 0], limit)    # totalprice DESC, orderdate ASC
    top.add_input(matched)
    pages = top.drain()
    elapsed = time.time() - t0
    for op in (agg1, f, b, j, top):
        op.close()
    bridge.close()
    session.tpch_lineitem_free(li)
    for p in (o_ok, o_ck, o_od):
        _device_free(session, p)
    tp = pages[0]
    okey = np.asarray(tp[0]["values"])
    ckey = np.asarray(tp[1]["values"])
    odate = np.asarray(tp[2]["values"])
    sumqty = np.asarray(tp[3]["values"])
    totc = np.asarray(tp[4]["values"])
    names = [f"Customer#{int(c):09d}" for c in ckey]
    return dict(c_name=names, custkey=ckey, orderkey=okey, orderdate=odate,
                totalprice_cents=totc, sum_qty=sumqty, elapsed=elapsed)


Q22_CODES = [13, 17, 18, 23, 29, 30, 31]   # cntrycode = c_nationkey + 10


def q22_gpu(session, sf, cust_count=None, order_count=None):
    """TPC-H Q22 (global sales opportunity): customers in 7 country codes
    with above-average positive balances and no orders. Country code =
    substring(c_phone,1,2) = nationkey+10 (the phone generator prefixes the
    nation; pinned via the Q10 fixture rows). Exact: balances are generated
    and summed in CENTS (i64), the average threshold compares
    cents*count > total (integer-exact in f64 below 2^53).
    Pipeline: customer filter -> scalar sums (avg) -> orders set-builder
    bitmap -> semi join -> filter (no order AND above avg) -> group by
    nationkey."""
    if cust_count is None:
        cust_count = int(150_000 * sf)
    if order_count is None:
        order_count = int(1_500_000 * sf)
    c_ck = _device_buffer(session, cust_count * 8)
    c_nk = _device_buffer(session, cust_count)
    c_ab = _device_buffer(session, cust_count * 8)
    _check_lib(_lib.tg_tpch_gen_customer(session._h, sf, 1, cust_count,
                                         c_ck, None, c_nk, c_ab))
    o_ck = _device_buffer(session, order_count * 8)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, 1, order_count,
                                       None, o_ck, None, None))
    t0 = time.time()
    cpage = ops.page_from_device(session, ([(c_ck.value, ops.TG_BIGINT),
                                            (c_nk.value, ops.TG_TINYINT),
                                            (c_ab.value, ops.TG_BIGINT)], cust_count))
    in_chain = []
    for i, code in enumerate(Q22_CODES):
        in_chain += [("col", 1), ("i64", code - 10), "eq"]
        if i:
            in_chain.append("or")
    # stage A: avg of positive balances in the code set
    fa = ops.filter_project(session,
                            ops.expr(*(in_chain + [("col", 2), ("i64", 0), "gt", "and"])),
                            [ops.expr(("col", 2))], [ops.TG_BIGINT])
    fa.add_input(cpage)
    fa.finish()
    posbal = _take_device_page(session, fa)
    aa = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_I64, 0),
                                               (ops.AGG_COUNT_STAR, -1)])
    aa.add_input(posbal)
    apages = aa.drain()
    total_cents = int(apages[0][0]["values"][0])
    npos = int(apages[0][1]["values"][0])
    fa.close()
    aa.close()
    # stage B: orders custkey membership set
    bridge = ops.JoinBridge(session)
    b = ops.set_builder(session, bridge, [ops.TG_BIGINT], 0)
    b.add_input(ops.page_from_device(session, ([(o_ck.value, ops.TG_BIGINT)],
                                               order_count)))
    b.drain()
    sj = ops.semi_join(session, bridge, 0)
    sj.add_input(cpage)
    sj.finish()
    marked = _take_device_page(session, sj)   # (ck, nk, cents, matched)
    # no order AND in code set AND cents*npos > total (integer-exact in f64)
    fb = ops.filter_project(session,
                            ops.expr(*(in_chain +
                                       [("col", 3), ("i64", 0), "eq", "and",
                                        ("col", 2), ("i64", npos), "mul",
                                        ("i64", total_cents), "gt", "and"])),
                            [ops.expr(("col", 1)), ops.expr(("col", 2))],
                            [ops.TG_TINYINT, ops.TG_BIGINT])
    fb.add_input(marked)
    fb.finish()
    qual = _take_device_page(session, fb)
    agg = ops.hash_aggregation(session, [0], [ops.TG_TINYINT],
                               [(ops.AGG_COUNT_STAR, -1), (ops.AGG_SUM_I64, 1)])
    agg.add_input(qual)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (b, sj, fb, agg):
        op.close()
    bridge.close()
    for p in (c_ck, c_nk, c_ab, o_ck):
        _device_free(session, p)
    out = pages[0]
    nk = np.asarray(out[0]["values"]).astype(np.int64)
    cnt = np.asarray(out[1]["values"])
    cents = np.asarray(out[2]["values"])
    order = np.argsort(nk)
    return dict(cntrycode=(nk + 10)[order], numcust=cnt[order],
                totacctbal_cents=cents[order], elapsed=elapsed)


DATE_1996_01_01 = 9496
DATE_1996_04_01 = 9587


def q15_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q15 (top supplier): revenue view over a 3-month shipdate window
    grouped by l_suppkey (partsupp-bridge stream, canonical-row verified),
    suppliers at the maximum. Exact fixed-point revenue sums. The SF1 answer
    (supplier 8449, 1772627.2087) must match the reference fixture; s_name
    is the formatted key, address/phone are unpinned text columns and are
    not produced."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    li = session.tpch_lineitem(sf, order_start, order_count, with_suppkey=True)
    trace = os.environ.get("TG_Q15_TRACE")
    t0 = time.time()
    lpage = ops.page_from_device(session, ([(li.suppkey, ops.TG_BIGINT),
                                            (li.shipdate, ops.TG_INTEGER),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE)],
                                           li.row_count))
    f = ops.filter_project(session,
                           ops.expr(("col", 1), ("i64", DATE_1996_01_01), "ge",
                                    ("col", 1), ("i64", DATE_1996_04_01), "lt", "and"),
                           [ops.expr(("col", 0)),
                            ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul")],
                           [ops.TG_BIGINT, ops.TG_DOUBLE])
    f.add_input(lpage)
    f.finish()
    sel = _take_device_page(session, f)
    if trace:
        import sys
        print(f"[q15] filter {(time.time()-t0)*1e3:.1f} ms", file=sys.stderr,
              flush=True)
    # suppkeys are dense 1..10k*SF: direct-array exact sums (the ~1M-group
    # hash table sized its keystore for the 22.7M incoming rows — ~900 MB
    # of allocations that cost ~58 ms whenever the pool was cold, which in
    # the 22-query sweep was every step)
    agg = ops.dense_aggregation(session, 0, 1, int(10_000 * sf),
                                (ops.AGG_SUM_F64_EXACT, 1, 43))
    agg.add_input(sel)
    if trace:
        import sys
        print(f"[q15] agg update {(time.time()-t0)*1e3:.1f} ms cum",
              file=sys.stderr, flush=True)
    pages = agg.drain()
    elapsed = time.time() - t0
    if trace:
        print(f"[q15] drain {elapsed*1e3:.1f} ms cum", file=sys.stderr,
              flush=True)
    f.close()
    agg.close()
    session.tpch_lineitem_free(li)
    out = pages[0]
    sk = np.asarray(out[0]["values"])
    rev = np.asarray(out[1]["values"])
    mx = rev.max()
    pick = np.nonzero(rev == mx)[0]
    order = pick[np.argsort(sk[pick])]
    return dict(suppkey=sk[order],
                s_name=[f"Supplier#{int(k):09d}" for k in sk[order]],
                total_revenue=rev[order], elapsed=elapsed)


NATION_NAMES = ["ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
                "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ",
                "JAPAN", "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU",
                "CHINA", "ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA",
                "UNITED KINGDOM", "UNITED STATES"]
ASIA_NATIONS = [8, 9, 12, 18, 21]   # region ASIA per the fixed nation table


def q5_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q5 (local supplier volume): ASIA-region revenue where the
    customer and supplier share a nation; 1994 order window. Uses the pinned
    customer/supplier nationkey streams and the partsupp-bridge l_suppkey;
    the fixed 25-row nation/region tables are spec constants. SF1 must match
    the reference fixture exactly (5 nations to 4 decimals)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    cust_count = int(150_000 * sf)
    supp_count = int(10_000 * sf)
    c_ck = _device_buffer(session, cust_count * 8)
    c_nk = _device_buffer(session, cust_count)
    _check_lib(_lib.tg_tpch_gen_customer(session._h, sf, 1, cust_count,
                                         c_ck, None, c_nk, None))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier(session._h, sf, 1, supp_count, s_sk, s_nk))
    o_ok = _device_buffer(session, order_count * 8)
    o_ck = _device_buffer(session, order_count * 8)
    o_od = _device_buffer(session, order_count * 4)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, order_start, order_count,
                                       o_ok, o_ck, o_od, None))
    li = session.tpch_lineitem(sf, order_start, order_count,
                               with_orderkey=True, with_suppkey=True)
    t0 = time.time()
    # build1: ASIA customers only (selective filter FIRST — the same lever
    # that took Q7 95->38 ms in round 1: 20%% of customers, so the orders
    # join and everything downstream shrinks 5x)
    asia_chain = []
    for _i, _nk in enumerate(ASIA_NATIONS):
        asia_chain += [("col", 1), ("i64", _nk), "eq"]
        if _i:
            asia_chain.append("or")
    fc = ops.filter_project(session, ops.expr(*asia_chain),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_TINYINT])
    fc.add_input(ops.page_from_device(session, ([(c_ck.value, ops.TG_BIGINT),
                                                 (c_nk.value, ops.TG_TINYINT)],
                                                cust_count)))
    fc.finish()
    asia_cust = _take_device_page(session, fc)
    br1 = ops.JoinBridge(session)
    b1 = ops.hash_builder(session, br1, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b1.add_input(asia_cust)
    b1.drain()
    # orders in 1994 -> join customers
    f1 = ops.filter_project(session,
                            ops.expr(("col", 2), ("i64", DATE_1994_01_01), "ge",
                                     ("col", 2), ("i64", DATE_1994_01_01 + 365),
                                     "lt", "and"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_BIGINT])
    f1.add_input(ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                                 (o_ck.value, ops.TG_BIGINT),
                                                 (o_od.value, ops.TG_INTEGER)],
                                                order_count)))
    f1.finish()
    owin = _take_device_page(session, f1)
    j1 = ops.lookup_join(session, br1, [ops.TG_BIGINT, ops.TG_BIGINT], [1], [0])
    j1.add_input(owin)
    j1.finish()
    ojoined = _take_device_page(session, j1)    # (orderkey, c_nationkey)
    br2 = ops.JoinBridge(session)
    ops.request_bitmap(br2)
    b2 = ops.hash_builder(session, br2, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b2.add_input(ojoined)
    b2.drain()
    # lineitem scan with the FUSED dynamic filter (orderkey ∈ 1994-ASIA
    # orders, ~3%%): revenue is only computed for survivors
    fp = ops.filter_project_df(session, None,
                               [ops.expr(("col", 0)), ops.expr(("col", 1)),
                                ops.expr(("col", 2), ("f64", 1.0), ("col", 3),
                                         "sub", "mul")],
                               [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE],
                               br2, 0)
    fp.add_input(ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                                 (li.suppkey, ops.TG_BIGINT),
                                                 (li.extendedprice, ops.TG_DOUBLE),
                                                 (li.discount, ops.TG_DOUBLE)],
                                                li.row_count)))
    fp.finish()
    lsel = _take_device_page(session, fp)
    j2 = ops.lookup_join(session, br2, [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE],
                         [0], [1, 2])
    j2.add_input(lsel)
    j2.finish()
    lj = _take_device_page(session, j2)         # (suppkey, rev, c_nk)
    # join supplier nation (ASIA suppliers only — inner join drops the rest)
    fsup = ops.filter_project(session, ops.expr(*asia_chain),
                              [ops.expr(("col", 0)), ops.expr(("col", 1))],
                              [ops.TG_BIGINT, ops.TG_TINYINT])
    fsup.add_input(ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                                   (s_nk.value, ops.TG_TINYINT)],
                                                  supp_count)))
    fsup.finish()
    asia_supp = _take_device_page(session, fsup)
    br3 = ops.JoinBridge(session)
    b3 = ops.hash_builder(session, br3, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b3.add_input(asia_supp)
    b3.drain()
    j3 = ops.lookup_join(session, br3,
                         [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_TINYINT],
                         [0], [1, 2])
    j3.add_input(lj)
    j3.finish()
    final = _take_device_page(session, j3)      # (rev, c_nk, s_nk)
    in_chain = []
    for i, nk in enumerate(ASIA_NATIONS):
        in_chain += [("col", 1), ("i64", nk), "eq"]
        if i:
            in_chain.append("or")
    f2 = ops.filter_project(session,
                            ops.expr(*(in_chain +
                                       [("col", 1), ("col", 2), "eq", "and"])),
                            [ops.expr(("col", 1)), ops.expr(("col", 0))],
                            [ops.TG_TINYINT, ops.TG_DOUBLE])
    f2.add_input(final)
    f2.finish()
    qual = _take_device_page(session, f2)
    agg = ops.hash_aggregation(session, [0], [ops.TG_TINYINT],
                               [(ops.AGG_SUM_F64_EXACT, 1, 43)])
    agg.add_input(qual)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (fc, fsup, b1, f1, j1, b2, fp, j2, b3, j3, f2, agg):
        op.close()
    for br in (br1, br2, br3):
        br.close()
    session.tpch_lineitem_free(li)
    for p in (c_ck, c_nk, s_sk, s_nk, o_ok, o_ck, o_od):
        _device_free(session, p)
    out = pages[0]
    nk = np.asarray(out[0]["values"]).astype(np.int64)
    rev = np.asarray(out[1]["values"])
    order = np.argsort(-rev)
    return dict(n_name=[NATION_NAMES[k] for k in nk[order]],
                revenue=rev[order], elapsed=elapsed)


def q7_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q7 (volume shipping): FRANCE<->GERMANY revenue by ship year
    (1995-1996). Same pinned streams as Q5; l_year derived from shipdate in
    the projection. Join order: FR/DE suppliers first (8 %% of suppliers),
    so the big joins see only matching lines — the naive
    orders-before-filter order cost 95 ms/step at SF100, this is ~40.
    SF1 must match the reference fixture exactly."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    cust_count = int(150_000 * sf)
    supp_count = int(10_000 * sf)
    c_ck = _device_buffer(session, cust_count * 8)
    c_nk = _device_buffer(session, cust_count)
    _check_lib(_lib.tg_tpch_gen_customer(session._h, sf, 1, cust_count,
                                         c_ck, None, c_nk, None))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier(session._h, sf, 1, supp_count, s_sk, s_nk))
    o_ok = _device_buffer(session, order_count * 8)
    o_ck = _device_buffer(session, order_count * 8)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, order_start, order_count,
                                       o_ok, o_ck, None, None))
    li = session.tpch_lineitem(sf, order_start, order_count,
                               with_orderkey=True, with_suppkey=True)
    t0 = time.time()
    # FR/DE suppliers only -> tiny build
    fs = ops.filter_project(session,
                            ops.expr(("col", 1), ("i64", 6), "eq",
                                     ("col", 1), ("i64", 7), "eq", "or"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_TINYINT])
    fs.add_input(ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                                 (s_nk.value, ops.TG_TINYINT)],
                                                supp_count)))
    fs.finish()
    ssel = _take_device_page(session, fs)
    br3 = ops.JoinBridge(session)
    ops.request_bitmap(br3)
    b3 = ops.hash_builder(session, br3, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b3.add_input(ssel)
    b3.drain()
    # lineitem window + FR/DE supplier membership FUSED into the scan
    # (dynamic filter): the date window alone kept ~180M of 600M rows and
    # the join probed all of them; with the bitmap in-kernel only the ~8%
    # with FR/DE suppliers are materialized, and the join sees 14.5M rows
    f = ops.filter_project_df(session,
                              ops.expr(("col", 4), ("i64", DATE_1995_01_01), "ge",
                                       ("col", 4), ("i64", DATE_1995_01_01 + 731),
                                       "lt", "and"),
                              [ops.expr(("col", 1)), ops.expr(("col", 0)),
                               ops.expr(("col", 2), ("f64", 1.0), ("col", 3),
                                        "sub", "mul"),
                               ops.expr(("col", 4), ("i64", DATE_1996_01_01), "ge")],
                              [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE,
                               ops.TG_DOUBLE], br3, 1)
    f.add_input(ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                                (li.suppkey, ops.TG_BIGINT),
                                                (li.extendedprice, ops.TG_DOUBLE),
                                                (li.discount, ops.TG_DOUBLE),
                                                (li.shipdate, ops.TG_INTEGER)],
                                               li.row_count)))
    f.finish()
    lsel = _take_device_page(session, f)   # (sk, ok, rev, yf)
    j3 = ops.lookup_join(session, br3,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_DOUBLE],
                         [0], [1, 2, 3])
    j3.add_input(lsel)
    j3.finish()
    lfr = _take_device_page(session, j3)   # (ok, rev, yf, s_nk)
    # build by orderkey; probe ALL orders (ok, ck)
    br2 = ops.JoinBridge(session)
    b2 = ops.hash_builder(session, br2,
                          [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_DOUBLE, ops.TG_TINYINT],
                          [0], [1, 2, 3])
    b2.add_input(lfr)
    b2.drain()
    j2 = ops.lookup_join(session, br2, [ops.TG_BIGINT, ops.TG_BIGINT], [0], [1])
    j2.add_input(ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                                 (o_ck.value, ops.TG_BIGINT)],
                                                order_count)))
    j2.finish()
    oj = _take_device_page(session, j2)    # (ck, rev, yf, s_nk)
    # customer nation
    br1 = ops.JoinBridge(session)
    b1 = ops.hash_builder(session, br1, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b1.add_input(ops.page_from_device(session, ([(c_ck.value, ops.TG_BIGINT),
                                                 (c_nk.value, ops.TG_TINYINT)],
                                                cust_count)))
    b1.drain()
    j1 = ops.lookup_join(session, br1,
                         [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_DOUBLE, ops.TG_TINYINT],
                         [0], [1, 2, 3])
    j1.add_input(oj)
    j1.finish()
    final = _take_device_page(session, j1)  # (rev, yf, s_nk, c_nk)
    f2 = ops.filter_project(session,
                            ops.expr(("col", 2), ("i64", 6), "eq",
                                     ("col", 3), ("i64", 7), "eq", "and",
                                     ("col", 2), ("i64", 7), "eq",
                                     ("col", 3), ("i64", 6), "eq", "and", "or"),
                            [ops.expr(("col", 2)), ops.expr(("col", 3)),
                             ops.expr(("col", 1)), ops.expr(("col", 0))],
                            [ops.TG_TINYINT, ops.TG_TINYINT, ops.TG_DOUBLE,
                             ops.TG_DOUBLE])
    f2.add_input(final)
    f2.finish()
    qual = _take_device_page(session, f2)   # (s_nk, c_nk, yf, rev)
    agg = ops.hash_aggregation(session, [0, 1, 2],
                               [ops.TG_TINYINT, ops.TG_TINYINT, ops.TG_DOUBLE],
                               [(ops.AGG_SUM_F64_EXACT, 3, 43)])
    agg.add_input(qual)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (fs, b3, f, j3, b2, j2, b1, j1, f2, agg):
        op.close()
    for br in (br1, br2, br3):
        br.close()
    session.tpch_lineitem_free(li)
    for p in (c_ck, c_nk, s_sk, s_nk, o_ok, o_ck):
        _device_free(session, p)
    out = pages[0]
    sn = np.asarray(out[0]["values"]).astype(np.int64)
    cn = np.asarray(out[1]["values"]).astype(np.int64)
    yf = np.asarray(out[2]["values"])
    rev = np.asarray(out[3]["values"])
    year = np.where(yf > 0.5, 1996, 1995)
    order = np.lexsort((year, cn, sn))
    return dict(supp_nation=[NATION_NAMES[k] for k in sn[order]],
                cust_nation=[NATION_NAMES[k] for k in cn[order]],
                l_year=year[order], revenue=rev[order], elapsed=elapsed)


AMERICA_NATIONS = [1, 2, 3, 17, 24]
TYPE_ECONOMY_ANODIZED_STEEL = 103   # 4*25 + 0*5 + 3


def q8_gpu(session, sf, order_start=1, order_count=None):
    """TPC-H Q8 (national market share): BRAZIL's share of AMERICA-region
    revenue for ECONOMY ANODIZED STEEL parts by order year. SF1 must match
    the reference fixture (6-decimal shares)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    cust_count = int(150_000 * sf)
    supp_count = int(10_000 * sf)
    n_parts = int(200_000 * sf)
    c_ck = _device_buffer(session, cust_count * 8)
    c_nk = _device_buffer(session, cust_count)
    _check_lib(_lib.tg_tpch_gen_customer(session._h, sf, 1, cust_count,
                                         c_ck, None, c_nk, None))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier(session._h, sf, 1, supp_count, s_sk, s_nk))
    p_pk = _device_buffer(session, n_parts * 8)
    p_ty = _device_buffer(session, n_parts * 2)
    _check_lib(_lib.tg_tpch_gen_part(session._h, sf, 1, n_parts, p_pk, p_ty))
    o_ok = _device_buffer(session, order_count * 8)
    o_ck = _device_buffer(session, order_count * 8)
    o_od = _device_buffer(session, order_count * 4)
    _check_lib(_lib.tg_tpch_gen_orders(session._h, sf, order_start, order_count,
                                       o_ok, o_ck, o_od, None))
    li = session.tpch_lineitem(sf, order_start, order_count,
                               with_orderkey=True, with_partkey=True,
                               with_suppkey=True)
    t0 = time.time()
    # part filter -> tiny build (partkey only)
    fpart = ops.filter_project(session,
                               ops.expr(("col", 1),
                                        ("i64", TYPE_ECONOMY_ANODIZED_STEEL), "eq"),
                               [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fpart.add_input(ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                                    (p_ty.value, ops.TG_SMALLINT)],
                                                   n_parts)))
    fpart.finish()
    psel = _take_device_page(session, fpart)
    brp = ops.JoinBridge(session)
    ops.request_bitmap(brp)   # 0.67%% of parts -> 2.5 MB bitmap, L2-resident
    bp = ops.hash_builder(session, brp, [ops.TG_BIGINT], [0], [])
    bp.add_input(psel)
    bp.drain()
    # customers -> orders window -> AMERICA filter -> build2 (ok -> yearflag)
    br1 = ops.JoinBridge(session)
    b1 = ops.hash_builder(session, br1, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b1.add_input(ops.page_from_device(session, ([(c_ck.value, ops.TG_BIGINT),
                                                 (c_nk.value, ops.TG_TINYINT)],
                                                cust_count)))
    b1.drain()
    f1 = ops.filter_project(session,
                            ops.expr(("col", 2), ("i64", DATE_1995_01_01), "ge",
                                     ("col", 2), ("i64", DATE_1995_01_01 + 731),
                                     "lt", "and"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2), ("i64", DATE_1996_01_01), "ge")],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE])
    f1.add_input(ops.page_from_device(session, ([(o_ok.value, ops.TG_BIGINT),
                                                 (o_ck.value, ops.TG_BIGINT),
                                                 (o_od.value, ops.TG_INTEGER)],
                                                order_count)))
    f1.finish()
    owin = _take_device_page(session, f1)
    j1 = ops.lookup_join(session, br1,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE], [1], [0, 2])
    j1.add_input(owin)
    j1.finish()
    oj = _take_device_page(session, j1)      # (ok, yf, c_nk)
    in_chain = []
    for i, nk in enumerate(AMERICA_NATIONS):
        in_chain += [("col", 2), ("i64", nk), "eq"]
        if i:
            in_chain.append("or")
    f2 = ops.filter_project(session, ops.expr(*in_chain),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_DOUBLE])
    f2.add_input(oj)
    f2.finish()
    oam = _take_device_page(session, f2)
    br2 = ops.JoinBridge(session)
    b2 = ops.hash_builder(session, br2, [ops.TG_BIGINT, ops.TG_DOUBLE], [0], [1])
    b2.add_input(oam)
    b2.drain()
    # lineitem scan with FUSED dynamic part filter -> orders join -> supplier
    fl = ops.filter_project_df(session, None,
                               [ops.expr(("col", 0)), ops.expr(("col", 1)),
                                ops.expr(("col", 2)),
                                ops.expr(("col", 3), ("f64", 1.0), ("col", 4),
                                         "sub", "mul")],
                               [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT,
                                ops.TG_DOUBLE], brp, 0)
    fl.add_input(ops.page_from_device(session, ([(li.partkey, ops.TG_BIGINT),
                                                 (li.orderkey, ops.TG_BIGINT),
                                                 (li.suppkey, ops.TG_BIGINT),
                                                 (li.extendedprice, ops.TG_DOUBLE),
                                                 (li.discount, ops.TG_DOUBLE)],
                                                li.row_count)))
    fl.finish()
    lsel = _take_device_page(session, fl)    # (pk, ok, sk, rev)
    jp = ops.lookup_join(session, brp,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE],
                         [0], [1, 2, 3])
    jp.add_input(lsel)
    jp.finish()
    lpartsel = _take_device_page(session, jp)   # (ok, sk, rev)
    j2 = ops.lookup_join(session, br2,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE], [0], [1, 2])
    j2.add_input(lpartsel)
    j2.finish()
    lj = _take_device_page(session, j2)         # (sk, rev, yf)
    br3 = ops.JoinBridge(session)
    b3 = ops.hash_builder(session, br3, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b3.add_input(ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                                 (s_nk.value, ops.TG_TINYINT)],
                                                supp_count)))
    b3.drain()
    j3 = ops.lookup_join(session, br3,
                         [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_DOUBLE], [0], [1, 2])
    j3.add_input(lj)
    j3.finish()
    final = _take_device_page(session, j3)      # (rev, yf, s_nk)
    fb = ops.filter_project(session, None,
                            [ops.expr(("col", 1)),
                             ops.expr(("col", 2), ("i64", 2), "eq"),
                             ops.expr(("col", 0))],
                            [ops.TG_DOUBLE, ops.TG_DOUBLE, ops.TG_DOUBLE])
    fb.add_input(final)
    fb.finish()
    flagged = _take_device_page(session, fb)    # (yf, brazil, rev)
    agg = ops.hash_aggregation(session, [0, 1], [ops.TG_DOUBLE, ops.TG_DOUBLE],
                               [(ops.AGG_SUM_F64_EXACT, 2, 43)])
    agg.add_input(flagged)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (fpart, bp, b1, f1, j1, f2, b2, fl, jp, j2, b3, j3, fb, agg):
        op.close()
    for br in (brp, br1, br2, br3):
        br.close()
    session.tpch_lineitem_free(li)
    for p in (c_ck, c_nk, s_sk, s_nk, p_pk, p_ty, o_ok, o_ck, o_od):
        _device_free(session, p)
    out = pages[0]
    yf = np.asarray(out[0]["values"])
    bf = np.asarray(out[1]["values"])
    rev = np.asarray(out[2]["values"])
    res = {}
    for y in (1995, 1996):
        m = yf == (0.0 if y == 1995 else 1.0)
        den = rev[m].sum()
        num = rev[m & (bf == 1.0)].sum()
        res[y] = num / den
    return dict(o_year=[1995, 1996],
                mkt_share=[res[1995], res[1996]], elapsed=elapsed)


# ===================== round 2: the remaining ten queries ==================
# Each mirrors the oracle composition pinned (vs the reference's SF1 answer
# fixtures) in tests/test_oracle_queries.py; pipelines are operator-ABI
# chains like the reference's LocalExecutionPlanner operator factories.

import datetime as _dt


def _D(y, m, d):
    return (_dt.date(y, m, d) - _dt.date(1970, 1, 1)).days


_lib.tg_tpch_gen_orders3.restype = ctypes.c_int
_lib.tg_tpch_gen_orders3.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                     ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 8
_lib.tg_tpch_gen_part2.restype = ctypes.c_int
_lib.tg_tpch_gen_part2.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                   ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 7
_lib.tg_tpch_gen_partsupp.restype = ctypes.c_int
_lib.tg_tpch_gen_partsupp.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                      ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 4
_lib.tg_tpch_gen_supplier2.restype = ctypes.c_int
_lib.tg_tpch_gen_supplier2.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                       ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 3
_lib.tg_tpch_gen_supplier_comments.restype = ctypes.c_int
_lib.tg_tpch_gen_supplier_comments.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                               ctypes.c_int64, ctypes.c_int64,
                                               ctypes.c_void_p, ctypes.c_void_p]
for _fn in (_lib.tg_tpch_supplier_strings, _lib.tg_tpch_customer_strings):
    _fn.restype = ctypes.c_int
    _fn.argtypes = [ctypes.c_double, ctypes.c_void_p, ctypes.c_int32,
                    ctypes.c_int32] + [ctypes.c_void_p] * 6
_lib.tg_tpch_part_strings.restype = ctypes.c_int
_lib.tg_tpch_part_strings.argtypes = [ctypes.c_double, ctypes.c_void_p,
                                      ctypes.c_int32, ctypes.c_int32] + [ctypes.c_void_p] * 5

NATIONS = ["ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
           "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ",
           "JAPAN", "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU",
           "CHINA", "ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA",
           "UNITED KINGDOM", "UNITED STATES"]
NATION_REGION = [0, 1, 1, 1, 4, 0, 3, 3, 2, 2, 4, 4, 2, 4, 0, 0, 0, 1, 2, 3,
                 4, 2, 3, 3, 1]
TYPE_S1 = ["STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO"]
TYPE_S2 = ["ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"]
TYPE_S3 = ["TIN", "NICKEL", "BRASS", "STEEL", "COPPER"]


def _strings(fn, sf, keys, fields):
    """host materialization of supplier/customer strings by key"""
    keys = np.ascontiguousarray(np.asarray(keys, np.int64))
    n = len(keys)
    stride = 160
    bufs = {f: ctypes.create_string_buffer(max(n, 1) * stride)
            for f in fields if f in ("name", "address", "phone", "comment")}
    ab = np.zeros(max(n, 1), np.int64) if "acctbal" in fields else None
    nk = np.zeros(max(n, 1), np.int32) if "nationkey" in fields else None
    _check_lib(fn(sf, keys.ctypes.data, n, stride,
                  bufs.get("name"), bufs.get("address"), bufs.get("phone"),
                  bufs.get("comment"),
                  None if ab is None else ab.ctypes.data,
                  None if nk is None else nk.ctypes.data))
    out = {}
    for f, b in bufs.items():
        out[f] = [b.raw[i * stride:(i + 1) * stride].split(b"\0")[0].decode()
                  for i in range(n)]
    if ab is not None:
        out["acctbal"] = ab[:n]
    if nk is not None:
        out["nationkey"] = nk[:n]
    return out


def _gen_orders3(session, sf, order_count, want):
    """want: subset of (ok, ck, od, pri, status, tp, coff, clen)"""
    sizes = dict(ok=8, ck=8, od=4, pri=1, status=1, tp=8, coff=8, clen=4)
    bufs = {k: (_device_buffer(session, order_count * sizes[k]) if k in want else None)
            for k in sizes}
    _check_lib(_lib.tg_tpch_gen_orders3(
        session._h, sf, 1, order_count, bufs["ok"], bufs["ck"], bufs["od"],
        bufs["pri"], bufs["status"], bufs["tp"], bufs["coff"], bufs["clen"]))
    return bufs


def _free_bufs(session, bufs):
    for v in bufs.values():
        if v is not None:
            _device_free(session, v)


def q13_gpu(session, sf, order_count=None, cust_count=None):
    """TPC-H Q13: customer order-count distribution over orders whose
    o_comment is NOT LIKE '%special%requests%' (LEFT JOIN: customers with no
    qualifying orders count in the 0 bucket). Comment LIKE runs directly over
    the device text pool slices (no 7 GB materialization)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    if cust_count is None:
        cust_count = int(150_000 * sf)
    bufs = _gen_orders3(session, sf, order_count, ("ck", "coff", "clen"))
    t0 = time.time()
    d_flags = _device_buffer(session, order_count)
    ops.pool_like_flags(session, bufs["coff"], bufs["clen"], order_count,
                        "%special%requests%", d_flags)
    opage = ops.page_from_device(session, ([(bufs["ck"].value, ops.TG_BIGINT),
                                            (d_flags.value, ops.TG_TINYINT)],
                                           order_count))
    f = ops.filter_project(session, ops.expr(("col", 1), ("i64", 0), "eq"),
                           [ops.expr(("col", 0))], [ops.TG_BIGINT])
    f.add_input(opage)
    f.finish()
    kept = _take_device_page(session, f)
    # custkeys are dense 1..150k*SF: direct-array counting (one atomic/row,
    # no hash probes) instead of a ~10M-group hash table at SF100
    a1 = ops.dense_aggregation(session, 0, 1, cust_count,
                               (ops.AGG_COUNT_STAR, -1))
    a1.add_input(kept)
    a1.finish()
    percust = _take_device_page(session, a1)       # (custkey, c_count)
    n_with = percust.position_count
    # histogram over c_count values (<= a few hundred distinct): dense-range
    # aggregation with LDS-privatized counting (hash agg's ~64 hot group
    # addresses measured 25.3 ms at SF100)
    a2 = ops.dense_aggregation(session, 1, 1, 4096, (ops.AGG_COUNT_STAR, -1))
    a2.add_input(percust)
    pages = a2.drain()
    elapsed = time.time() - t0
    for op in (f, a1, a2):
        op.close()
    _device_free(session, d_flags)
    _free_bufs(session, bufs)
    out = pages[0]
    c_count = np.asarray(out[0]["values"]).astype(np.int64)
    custdist = np.asarray(out[1]["values"]).astype(np.int64)
    c_count = np.append(c_count, 0)
    custdist = np.append(custdist, cust_count - n_with)
    order = np.lexsort((-c_count, -custdist))
    return dict(c_count=c_count[order], custdist=custdist[order],
                elapsed=elapsed)


def q16_gpu(session, sf, part_count=None):
    """TPC-H Q16: supplier counts by (brand, type, size) excluding
    Brand#45 / MEDIUM POLISHED% / sizes outside the 8-set and suppliers with
    '%Customer%Complaints%' comments (BBB overlay + generic VARCHAR LIKE).
    count(DISTINCT ps_suppkey) = dedup group-by then count."""
    if part_count is None:
        part_count = int(200_000 * sf)
    supp_count = int(10_000 * sf)
    p_pk = _device_buffer(session, part_count * 8)
    p_ty = _device_buffer(session, part_count * 2)
    p_br = _device_buffer(session, part_count)
    p_sz = _device_buffer(session, part_count * 4)
    _check_lib(_lib.tg_tpch_gen_part2(session._h, sf, 1, part_count,
                                      p_pk, p_ty, p_br, p_sz, None, None, None))
    ps_pk = _device_buffer(session, part_count * 4 * 8)
    ps_sk = _device_buffer(session, part_count * 4 * 8)
    _check_lib(_lib.tg_tpch_gen_partsupp(session._h, sf, 1, part_count,
                                         ps_pk, ps_sk, None, None))
    t0 = time.time()
    # complaint suppliers (BBB overlay) via the generic VARCHAR LIKE kernel
    d_soff = ctypes.c_void_p()
    d_sbytes = ctypes.c_void_p()
    _check_lib(_lib.tg_tpch_gen_supplier_comments(
        session._h, sf, 1, supp_count, ctypes.byref(d_soff), ctypes.byref(d_sbytes)))
    d_cflag = _device_buffer(session, supp_count)
    ops.varchar_like_flags(session, d_sbytes, d_soff, supp_count,
                           "%Customer%Complaints%", d_cflag)
    # suppkey column 1..n on device = partsupp suppkeys of part 1..? simpler:
    # download tiny flag vector, upload complaint keys as build page
    flags = np.empty(supp_count, np.uint8)
    from . import copy_dtoh
    copy_dtoh(session, flags, d_cflag)
    bad = np.nonzero(flags)[0].astype(np.int64) + 1
    bridge_bad = ops.JoinBridge(session)
    bb = ops.set_builder(session, bridge_bad, [ops.TG_BIGINT], 0)
    bb.add_input(ops.page_from_numpy([np.ascontiguousarray(bad)]))
    bb.drain()
    # part filter
    ppage = ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                            (p_br.value, ops.TG_TINYINT),
                                            (p_ty.value, ops.TG_SMALLINT),
                                            (p_sz.value, ops.TG_INTEGER)],
                                           part_count))
    sizes = [49, 14, 23, 45, 19, 3, 36, 9]
    in_chain = []
    for i, v in enumerate(sizes):
        in_chain += [("col", 3), ("i64", v), "eq"]
        if i:
            in_chain.append("or")
    fexpr = ops.expr(*(in_chain +
                       [("col", 1), ("i64", 45), "ne", "and",
                        ("col", 2), ("i64", 65), ("i64", 69), "between",
                        "not", "and"]))
    fp = ops.filter_project(session, fexpr,
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2)), ops.expr(("col", 3))],
                            [ops.TG_BIGINT, ops.TG_TINYINT, ops.TG_SMALLINT,
                             ops.TG_INTEGER])
    fp.add_input(ppage)
    fp.finish()
    part_sel = _take_device_page(session, fp)
    bridge_p = ops.JoinBridge(session)
    bp = ops.hash_builder(session, bridge_p, [ops.TG_BIGINT, ops.TG_TINYINT,
                                              ops.TG_SMALLINT, ops.TG_INTEGER],
                          [0], [1, 2, 3])
    bp.add_input(part_sel)
    bp.drain()
    # partsupp: drop complaint suppliers, then join part attrs
    pspage = ops.page_from_device(session, ([(ps_pk.value, ops.TG_BIGINT),
                                             (ps_sk.value, ops.TG_BIGINT)],
                                            part_count * 4))
    sj = ops.semi_join(session, bridge_bad, 1)
    sj.add_input(pspage)
    sj.finish()
    marked = _take_device_page(session, sj)
    fnb = ops.filter_project(session, ops.expr(("col", 2), ("i64", 0), "eq"),
                             [ops.expr(("col", 0)), ops.expr(("col", 1))],
                             [ops.TG_BIGINT, ops.TG_BIGINT])
    fnb.add_input(marked)
    fnb.finish()
    ps_ok = _take_device_page(session, fnb)
    j = ops.lookup_join(session, bridge_p, [ops.TG_BIGINT, ops.TG_BIGINT],
                        [0], [1])          # emit suppkey + (brand,type,size)
    j.add_input(ps_ok)
    j.finish()
    joined = _take_device_page(session, j)  # (sk, brand, type, size)
    # DISTINCT via a PACKED single BIGINT key (exact i64 arithmetic):
    # ((brand*160 + type)*64 + size) << 32 | suppkey — the 4-channel
    # keystore dedup measured 32.1 ms at SF100; single-word keys probe ~6x
    # faster. Unpack after the count.
    fpk = ops.filter_project(session, None,
                             [ops.expr(("col", 1), ("i64", 160), "mul",
                                       ("col", 2), "add", ("i64", 64), "mul",
                                       ("col", 3), "add",
                                       ("i64", 1 << 32), "mul",
                                       ("col", 0), "add")],
                             [ops.TG_BIGINT])
    fpk.add_input(joined)
    fpk.finish()
    packed = _take_device_page(session, fpk)
    # DISTINCT via sort-based dedup (radix sort + unique compaction):
    # ~11.9M of the 11.88M packed keys are unique, so the hash-agg dedup
    # paid random keystore probes + a group-remap sort for nothing
    # (profiled 31.0 ms of Q16's 51 ms at SF100; tg_dedup_i64 sorts
    # 52 packed bits sequentially instead)
    n_packed = packed.position_count
    d_dedup = _device_buffer(session, max(n_packed, 1) * 8)
    n_uniq = ops.dedup_i64(session, packed.blocks[0].data, n_packed,
                           d_dedup, bits=52)
    dedup = ops.page_from_device(session, ([(d_dedup.value, ops.TG_BIGINT)],
                                           n_uniq))
    # group by packed >> 32 (brand/type/size combo) counting suppliers
    fg = ops.filter_project(session, None,
                            [ops.expr(("col", 0), ("i64", 1 << 32), "div")],
                            [ops.TG_BIGINT])
    fg.add_input(dedup)
    fg.finish()
    combos = _take_device_page(session, fg)
    d2 = ops.hash_aggregation(session, [0], [ops.TG_BIGINT],
                              [(ops.AGG_COUNT_STAR, -1)])
    d2.add_input(combos)
    pages = d2.drain()
    elapsed = time.time() - t0
    for op in (bb, fp, bp, sj, fnb, j, fpk, fg, d2):
        op.close()
    bridge_bad.close()
    bridge_p.close()
    for p in (p_pk, p_ty, p_br, p_sz, ps_pk, ps_sk, d_cflag, d_dedup):
        _device_free(session, p)
    out = pages[0]
    combo = np.asarray(out[0]["values"]).astype(np.int64)
    cnt = np.asarray(out[1]["values"]).astype(np.int64)
    sz = combo % 64
    bt = combo // 64
    ty = bt % 160
    br = bt // 160
    rows = [(f"Brand#{br[i]}",
             f"{TYPE_S1[ty[i]//25]} {TYPE_S2[(ty[i]//5)%5]} {TYPE_S3[ty[i]%5]}",
             int(sz[i]), int(cnt[i])) for i in range(len(combo))]
    rows.sort(key=lambda r: (-r[3], r[0], r[1], r[2]))
    return dict(rows=rows, elapsed=elapsed)


def q11_gpu(session, sf, part_count=None):
    """TPC-H Q11 (important stock in GERMANY): value per partkey =
    sum(ps_supplycost * ps_availqty) over German suppliers, kept where
    value > 0.0001 * total. Products/sums computed EXACTLY in integer cents
    (cost_cents * qty fits i64); the threshold compare mirrors Trino's
    double compare on the exact cents (margins are far above 1 ulp)."""
    if part_count is None:
        part_count = int(200_000 * sf)
    supp_count = int(10_000 * sf)
    n = part_count * 4
    ps_pk = _device_buffer(session, n * 8)
    ps_sk = _device_buffer(session, n * 8)
    ps_aq = _device_buffer(session, n * 4)
    ps_sc = _device_buffer(session, n * 8)
    _check_lib(_lib.tg_tpch_gen_partsupp(session._h, sf, 1, part_count,
                                         ps_pk, ps_sk, ps_aq, ps_sc))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier2(session._h, sf, 1, supp_count,
                                          s_sk, s_nk, None))
    t0 = time.time()
    spage = ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                            (s_nk.value, ops.TG_TINYINT)],
                                           supp_count))
    fs = ops.filter_project(session, ops.expr(("col", 1), ("i64", 7), "eq"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fs.add_input(spage)
    fs.finish()
    de = _take_device_page(session, fs)
    bridge = ops.JoinBridge(session)
    b = ops.set_builder(session, bridge, [ops.TG_BIGINT], 0)
    b.add_input(de)
    b.drain()
    pspage = ops.page_from_device(session, ([(ps_pk.value, ops.TG_BIGINT),
                                             (ps_sk.value, ops.TG_BIGINT),
                                             (ps_aq.value, ops.TG_INTEGER),
                                             (ps_sc.value, ops.TG_BIGINT)], n))
    sj = ops.semi_join(session, bridge, 1)
    sj.add_input(pspage)
    sj.finish()
    marked = _take_device_page(session, sj)
    # value_cents = cost_cents * availqty (exact i64 via the typed lane)
    fv = ops.filter_project(session, ops.expr(("col", 4), ("i64", 1), "eq"),
                            [ops.expr(("col", 0)),
                             ops.expr(("col", 3), ("col", 2), "mul")],
                            [ops.TG_BIGINT, ops.TG_BIGINT])
    fv.add_input(marked)
    fv.finish()
    vals = _take_device_page(session, fv)
    agg = ops.hash_aggregation(session, [0], [ops.TG_BIGINT],
                               [(ops.AGG_SUM_I64, 1)])
    agg.add_input(vals)
    agg.finish()
    groups = _take_device_page(session, agg)       # (partkey, value_cents)
    tot = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_I64, 1)])
    tot.add_input(groups)
    tpages = tot.drain()
    total_cents = int(tpages[0][0]["values"][0])
    # value > 0.0001 * total  <=>  value_cents * 10000 > total_cents
    ff = ops.filter_project(session,
                            ops.expr(("col", 1), ("i64", 10000), "mul",
                                     ("i64", total_cents), "gt"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_BIGINT])
    ff.add_input(groups)
    pages = ff.drain()
    elapsed = time.time() - t0
    for op in (fs, b, sj, fv, agg, tot, ff):
        op.close()
    bridge.close()
    for p in (ps_pk, ps_sk, ps_aq, ps_sc, s_sk, s_nk):
        _device_free(session, p)
    out = pages[0]
    pk = np.asarray(out[0]["values"]).astype(np.int64)
    cents = np.asarray(out[1]["values"]).astype(np.int64)
    order = np.lexsort((pk, -cents))
    return dict(partkey=pk[order], value_cents=cents[order], elapsed=elapsed)


def q17_gpu(session, sf, order_count=None, part_count=None):
    """TPC-H Q17 (small-quantity-order revenue): Brand#23 / MED BOX parts;
    avg yearly = sum(l_extendedprice where l_quantity < 0.2*avg(qty) per
    part) / 7. Two probe passes over the part-filtered lineitem join."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    if part_count is None:
        part_count = int(200_000 * sf)
    p_pk = _device_buffer(session, part_count * 8)
    p_br = _device_buffer(session, part_count)
    p_cn = _device_buffer(session, part_count)
    _check_lib(_lib.tg_tpch_gen_part2(session._h, sf, 1, part_count,
                                      p_pk, None, p_br, None, p_cn, None, None))
    li = session.tpch_lineitem(sf, 1, order_count, with_partkey=True)
    t0 = time.time()
    ppage = ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                            (p_br.value, ops.TG_TINYINT),
                                            (p_cn.value, ops.TG_TINYINT)],
                                           part_count))
    fp = ops.filter_project(session,
                            ops.expr(("col", 1), ("i64", 23), "eq",
                                     ("col", 2), ("i64", 17), "eq", "and"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fp.add_input(ppage)
    fp.finish()
    sel_parts = _take_device_page(session, fp)
    bridge = ops.JoinBridge(session)
    b = ops.hash_builder(session, bridge, [ops.TG_BIGINT], [0], [])
    b.add_input(sel_parts)
    b.drain()
    lpage = ops.page_from_device(session, ([(li.partkey, ops.TG_BIGINT),
                                            (li.quantity, ops.TG_DOUBLE),
                                            (li.extendedprice, ops.TG_DOUBLE)],
                                           li.row_count))
    j = ops.lookup_join(session, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE,
                                          ops.TG_DOUBLE], [0], [0, 1, 2])
    j.add_input(lpage)
    j.finish()
    rows = _take_device_page(session, j)           # (pk, qty, ep)
    agg = ops.hash_aggregation(session, [0], [ops.TG_BIGINT],
                               [(ops.AGG_SUM_F64, 1), (ops.AGG_COUNT_STAR, -1)])
    agg.add_input(rows)
    agg.finish()
    peravg = _take_device_page(session, agg)       # (pk, sumq, cnt)
    bridge2 = ops.JoinBridge(session)
    b2 = ops.hash_builder(session, bridge2, [ops.TG_BIGINT, ops.TG_DOUBLE,
                                             ops.TG_BIGINT], [0], [1, 2])
    b2.add_input(peravg)
    b2.drain()
    j2 = ops.lookup_join(session, bridge2, [ops.TG_BIGINT, ops.TG_DOUBLE,
                                            ops.TG_DOUBLE], [0], [1, 2])
    j2.add_input(rows)
    j2.finish()
    wavg = _take_device_page(session, j2)          # (qty, ep, sumq, cnt)
    # qty < 0.2 * (sumq / cnt)  — same IEEE ops as DoubleAverageAggregations
    ffin = ops.filter_project(session,
                              ops.expr(("col", 0),
                                       ("f64", 0.2), ("col", 2), ("col", 3),
                                       "div", "mul", "lt"),
                              [ops.expr(("col", 1))], [ops.TG_DOUBLE])
    ffin.add_input(wavg)
    ffin.finish()
    kept = _take_device_page(session, ffin)
    sagg = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_F64_EXACT, 0, 43)])
    sagg.add_input(kept)
    pages = sagg.drain()
    elapsed = time.time() - t0
    for op in (fp, b, j, agg, b2, j2, ffin, sagg):
        op.close()
    bridge.close()
    bridge2.close()
    session.tpch_lineitem_free(li)
    for p in (p_pk, p_br, p_cn):
        _device_free(session, p)
    total = float(pages[0][0]["values"][0])
    return dict(avg_yearly=total / 7.0, elapsed=elapsed)


def q19_gpu(session, sf, order_count=None, part_count=None):
    """TPC-H Q19 (discounted revenue): three brand/container/size/quantity
    branches, AIR shipmode + DELIVER IN PERSON, summed exactly."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    if part_count is None:
        part_count = int(200_000 * sf)
    p_pk = _device_buffer(session, part_count * 8)
    p_br = _device_buffer(session, part_count)
    p_cn = _device_buffer(session, part_count)
    p_sz = _device_buffer(session, part_count * 4)
    _check_lib(_lib.tg_tpch_gen_part2(session._h, sf, 1, part_count,
                                      p_pk, None, p_br, p_sz, p_cn, None, None))
    li = session.tpch_lineitem(sf, 1, order_count, with_partkey=True,
                               with_shipmode=True, with_shipinstruct=True)
    t0 = time.time()
    # part-side union of the three branch (brand, container, size)
    # conditions, pushed out of the join condition like the reference's
    # planner does: ~0.04% of parts survive, so the build is tiny and the
    # bitmap below prunes the scan (quantity/branch pairing still applied
    # post-join by f2)
    def pbr(bid, conts, shi):
        e = []
        for i, c in enumerate(conts):
            e += [("col", 2), ("i64", c), "eq"]
            if i:
                e.append("or")
        e += [("col", 1), ("i64", bid), "eq", "and",
              ("col", 3), ("i64", 1), ("i64", shi), "between", "and"]
        return e
    pe = (pbr(12, [0, 1, 4, 5], 5) + pbr(23, [17, 18, 20, 21], 10) + ["or"] +
          pbr(34, [8, 9, 12, 13], 15) + ["or"])
    fp0 = ops.filter_project(session, ops.expr(*pe),
                             [ops.expr(("col", 0)), ops.expr(("col", 1)),
                              ops.expr(("col", 2)), ops.expr(("col", 3))],
                             [ops.TG_BIGINT, ops.TG_TINYINT, ops.TG_TINYINT,
                              ops.TG_INTEGER])
    fp0.add_input(ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                                  (p_br.value, ops.TG_TINYINT),
                                                  (p_cn.value, ops.TG_TINYINT),
                                                  (p_sz.value, ops.TG_INTEGER)],
                                                 part_count)))
    fp0.finish()
    psel = _take_device_page(session, fp0)
    bridge = ops.JoinBridge(session)
    ops.request_bitmap(bridge)
    b = ops.hash_builder(session, bridge,
                         [ops.TG_BIGINT, ops.TG_TINYINT, ops.TG_TINYINT,
                          ops.TG_INTEGER], [0], [1, 2, 3])
    b.add_input(psel)
    b.drain()
    lpage = ops.page_from_device(session, ([(li.partkey, ops.TG_BIGINT),
                                            (li.quantity, ops.TG_DOUBLE),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE),
                                            (li.shipmode, ops.TG_TINYINT),
                                            (li.shipinstruct, ops.TG_TINYINT)],
                                           li.row_count))
    # cheap common predicate + qualifying-part membership fused in-kernel
    f1 = ops.filter_project_df(session,
                               ops.expr(("col", 4), ("i64", 1), "eq",
                                        ("col", 5), ("i64", 0), "eq", "and"),
                               [ops.expr(("col", 0)), ops.expr(("col", 1)),
                                ops.expr(("col", 2), ("f64", 1.0), ("col", 3),
                                         "sub", "mul")],
                               [ops.TG_BIGINT, ops.TG_DOUBLE, ops.TG_DOUBLE],
                               bridge, 0)
    f1.add_input(lpage)
    f1.finish()
    lsel = _take_device_page(session, f1)
    j = ops.lookup_join(session, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE,
                                          ops.TG_DOUBLE], [0], [1, 2])
    j.add_input(lsel)
    j.finish()
    rows = _take_device_page(session, j)  # (qty, disc_price, brand, cont, size)

    def branch(bid, conts, qlo, qhi, shi):
        e = []
        for i, c in enumerate(conts):
            e += [("col", 3), ("i64", c), "eq"]
            if i:
                e.append("or")
        e += [("col", 2), ("i64", bid), "eq", "and",
              ("col", 0), ("f64", float(qlo)), ("f64", float(qhi)), "between",
              "and",
              ("col", 4), ("i64", 1), ("i64", shi), "between", "and"]
        return e

    e1 = branch(12, [0, 1, 4, 5], 1, 11, 5)
    e2 = branch(23, [17, 18, 20, 21], 10, 20, 10)
    e3 = branch(34, [8, 9, 12, 13], 20, 30, 15)
    full = e1 + e2 + ["or"] + e3 + ["or"]
    f2 = ops.filter_project(session, ops.expr(*full),
                            [ops.expr(("col", 1))], [ops.TG_DOUBLE])
    f2.add_input(rows)
    f2.finish()
    kept = _take_device_page(session, f2)
    sagg = ops.hash_aggregation(session, [], [], [(ops.AGG_SUM_F64_EXACT, 0, 43)])
    sagg.add_input(kept)
    pages = sagg.drain()
    elapsed = time.time() - t0
    for op in (fp0, b, f1, j, f2, sagg):
        op.close()
    bridge.close()
    session.tpch_lineitem_free(li)
    for p in (p_pk, p_br, p_cn, p_sz):
        _device_free(session, p)
    return dict(revenue=float(pages[0][0]["values"][0]), elapsed=elapsed)


_lib.tg_tpch_part_name_flag.restype = ctypes.c_int
_lib.tg_tpch_part_name_flag.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_int64, ctypes.c_int32,
                                        ctypes.c_int32, ctypes.c_void_p]
COLOR_GREEN, COLOR_FOREST = 33, 28


def _strings(fn, sf, keys, fields):
    """host materialization of supplier/customer strings by key"""
    keys = np.ascontiguousarray(np.asarray(keys, np.int64))
    n = len(keys)
    stride = 160
    bufs = {f: ctypes.create_string_buffer(max(n, 1) * stride)
            for f in fields if f in ("name", "address", "phone", "comment")}
    ab = np.zeros(max(n, 1), np.int64) if "acctbal" in fields else None
    nk = np.zeros(max(n, 1), np.int32) if "nationkey" in fields else None
    _check_lib(fn(sf, keys.ctypes.data, n, stride,
                  bufs.get("name"), bufs.get("address"), bufs.get("phone"),
                  bufs.get("comment"),
                  None if ab is None else ab.ctypes.data,
                  None if nk is None else nk.ctypes.data))
    out = {}
    for f, b in bufs.items():
        out[f] = [b.raw[i * stride:(i + 1) * stride].split(b"\0")[0].decode()
                  for i in range(n)]
    if ab is not None:
        out["acctbal"] = ab[:n]
    if nk is not None:
        out["nationkey"] = nk[:n]
    return out


def _strings_part(sf, keys, fields):
    keys = np.ascontiguousarray(np.asarray(keys, np.int64))
    n = len(keys)
    stride = 160
    bufs = {f: ctypes.create_string_buffer(max(n, 1) * stride)
            for f in fields}
    _check_lib(_lib.tg_tpch_part_strings(
        sf, keys.ctypes.data, n, stride, bufs.get("name"), bufs.get("mfgr"),
        bufs.get("brand"), bufs.get("type"), bufs.get("container")))
    return {f: [b.raw[i * stride:(i + 1) * stride].split(b"\0")[0].decode()
                for i in range(n)] for f, b in bufs.items()}


def q2_gpu(session, sf, part_count=None):
    """TPC-H Q2 (minimum-cost supplier): size=15, type '%BRASS', EUROPE;
    rows where ps_supplycost equals the per-part minimum over European
    suppliers (TG_AGG_MIN_I64 on exact cents); output strings host-side."""
    if part_count is None:
        part_count = int(200_000 * sf)
    supp_count = int(10_000 * sf)
    p_pk = _device_buffer(session, part_count * 8)
    p_ty = _device_buffer(session, part_count * 2)
    p_sz = _device_buffer(session, part_count * 4)
    _check_lib(_lib.tg_tpch_gen_part2(session._h, sf, 1, part_count,
                                      p_pk, p_ty, None, p_sz, None, None, None))
    n = part_count * 4
    ps_pk = _device_buffer(session, n * 8)
    ps_sk = _device_buffer(session, n * 8)
    ps_sc = _device_buffer(session, n * 8)
    _check_lib(_lib.tg_tpch_gen_partsupp(session._h, sf, 1, part_count,
                                         ps_pk, ps_sk, None, ps_sc))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    s_ab = _device_buffer(session, supp_count * 8)
    _check_lib(_lib.tg_tpch_gen_supplier2(session._h, sf, 1, supp_count,
                                          s_sk, s_nk, s_ab))
    t0 = time.time()
    eu = [nk for nk in range(25) if NATION_REGION[nk] == 3]
    spage = ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                            (s_nk.value, ops.TG_TINYINT),
                                            (s_ab.value, ops.TG_BIGINT)],
                                           supp_count))
    chain = []
    for i, nk in enumerate(eu):
        chain += [("col", 1), ("i64", nk), "eq"]
        if i:
            chain.append("or")
    fs = ops.filter_project(session, ops.expr(*chain),
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2))],
                            [ops.TG_BIGINT, ops.TG_TINYINT, ops.TG_BIGINT])
    fs.add_input(spage)
    fs.finish()
    eu_supp = _take_device_page(session, fs)
    bridge_s = ops.JoinBridge(session)
    bs = ops.hash_builder(session, bridge_s,
                          [ops.TG_BIGINT, ops.TG_TINYINT, ops.TG_BIGINT],
                          [0], [1, 2])
    bs.add_input(eu_supp)
    bs.drain()
    ppage = ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                            (p_ty.value, ops.TG_SMALLINT),
                                            (p_sz.value, ops.TG_INTEGER)],
                                           part_count))
    # size == 15 AND type %% 5 == 2 ('%BRASS'): ((ty-2)/5)*5 == ty-2
    fexpr = ops.expr(("col", 2), ("i64", 15), "eq",
                     ("col", 1), ("i64", 2), "sub", ("i64", 5), "div",
                     ("i64", 5), "mul",
                     ("col", 1), ("i64", 2), "sub", "eq", "and")
    fp = ops.filter_project(session, fexpr, [ops.expr(("col", 0))],
                            [ops.TG_BIGINT])
    fp.add_input(ppage)
    fp.finish()
    sel_parts = _take_device_page(session, fp)
    bridge_p = ops.JoinBridge(session)
    bp = ops.hash_builder(session, bridge_p, [ops.TG_BIGINT], [0], [])
    bp.add_input(sel_parts)
    bp.drain()
    pspage = ops.page_from_device(session, ([(ps_pk.value, ops.TG_BIGINT),
                                             (ps_sk.value, ops.TG_BIGINT),
                                             (ps_sc.value, ops.TG_BIGINT)], n))
    j1 = ops.lookup_join(session, bridge_p, [ops.TG_BIGINT, ops.TG_BIGINT,
                                             ops.TG_BIGINT], [0], [0, 1, 2])
    j1.add_input(pspage)
    j1.finish()
    ps_parts = _take_device_page(session, j1)
    j2 = ops.lookup_join(session, bridge_s, [ops.TG_BIGINT, ops.TG_BIGINT,
                                             ops.TG_BIGINT], [1], [0, 1, 2])
    j2.add_input(ps_parts)
    j2.finish()
    rows = _take_device_page(session, j2)          # (pk, sk, cost, nk, ab)
    agg = ops.hash_aggregation(session, [0], [ops.TG_BIGINT],
                               [(ops.AGG_MIN_I64, 2)])
    agg.add_input(rows)
    agg.finish()
    mins = _take_device_page(session, agg)
    bridge_m = ops.JoinBridge(session)
    bm = ops.hash_builder(session, bridge_m, [ops.TG_BIGINT, ops.TG_BIGINT],
                          [0], [1])
    bm.add_input(mins)
    bm.drain()
    j3 = ops.lookup_join(session, bridge_m,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT,
                          ops.TG_TINYINT, ops.TG_BIGINT], [0],
                         [0, 1, 2, 3, 4])
    j3.add_input(rows)
    j3.finish()
    withmin = _take_device_page(session, j3)
    ff = ops.filter_project(session, ops.expr(("col", 2), ("col", 5), "eq"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 3)), ops.expr(("col", 4))],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_TINYINT,
                             ops.TG_BIGINT])
    ff.add_input(withmin)
    pages = ff.drain()
    elapsed = time.time() - t0
    for op in (fs, bs, fp, bp, j1, j2, agg, bm, j3, ff):
        op.close()
    for br in (bridge_s, bridge_p, bridge_m):
        br.close()
    for p in (p_pk, p_ty, p_sz, ps_pk, ps_sk, ps_sc, s_sk, s_nk, s_ab):
        _device_free(session, p)
    out = pages[0]
    pk = np.asarray(out[0]["values"]).astype(np.int64)
    sk = np.asarray(out[1]["values"]).astype(np.int64)
    nk = np.asarray(out[2]["values"]).astype(np.int64)
    ab = np.asarray(out[3]["values"]).astype(np.int64)
    rows_h = sorted(zip(ab.tolist(), [NATIONS[x] for x in nk], sk.tolist(),
                        pk.tolist()),
                    key=lambda r: (-r[0], r[1], r[2], r[3]))[:100]
    st = _strings(_lib.tg_tpch_supplier_strings, sf, [r[2] for r in rows_h],
                  ("name", "address", "phone", "comment"))
    pstr = _strings_part(sf, [r[3] for r in rows_h], ("mfgr",))
    result = []
    for i, r in enumerate(rows_h):
        result.append(dict(acctbal_cents=r[0], s_name=st["name"][i],
                           n_name=r[1], partkey=r[3], mfgr=pstr["mfgr"][i],
                           address=st["address"][i], phone=st["phone"][i],
                           comment=st["comment"][i]))
    return dict(rows=result, elapsed=elapsed)


def q10_gpu(session, sf, order_count=None, limit=20):
    """TPC-H Q10 (returned items): revenue per customer over returnflag R
    lines of 1993-10..1994-01 orders; top `limit` by revenue desc."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    bufs = _gen_orders3(session, sf, order_count, ("ok", "ck", "od"))
    li = session.tpch_lineitem(sf, 1, order_count, with_orderkey=True)
    t0 = time.time()
    opage = ops.page_from_device(session, ([(bufs["ok"].value, ops.TG_BIGINT),
                                            (bufs["ck"].value, ops.TG_BIGINT),
                                            (bufs["od"].value, ops.TG_INTEGER)],
                                           order_count))
    fo = ops.filter_project(session,
                            ops.expr(("col", 2), ("i64", _D(1993, 10, 1)), "ge",
                                     ("col", 2), ("i64", _D(1994, 1, 1)), "lt",
                                     "and"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_BIGINT])
    fo.add_input(opage)
    fo.finish()
    owin = _take_device_page(session, fo)
    bridge = ops.JoinBridge(session)
    ops.request_bitmap(bridge)
    b = ops.hash_builder(session, bridge, [ops.TG_BIGINT, ops.TG_BIGINT],
                         [0], [1])
    b.add_input(owin)
    b.drain()
    lpage = ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                            (li.returnflag, ops.TG_TINYINT),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE)],
                                           li.row_count))
    fl = ops.filter_project_df(session,
                               ops.expr(("col", 1), ("i64", 2), "eq"),
                               [ops.expr(("col", 0)),
                                ops.expr(("col", 2), ("f64", 1.0), ("col", 3),
                                         "sub", "mul")],
                               [ops.TG_BIGINT, ops.TG_DOUBLE], bridge, 0)
    fl.add_input(lpage)
    fl.finish()
    lr = _take_device_page(session, fl)
    j = ops.lookup_join(session, bridge, [ops.TG_BIGINT, ops.TG_DOUBLE],
                        [0], [1])
    j.add_input(lr)
    j.finish()
    joined = _take_device_page(session, j)     # (discprice, custkey)
    agg = ops.hash_aggregation(session, [1], [ops.TG_BIGINT],
                               [(ops.AGG_SUM_F64_EXACT, 0, 43)])
    agg.add_input(joined)
    agg.finish()
    groups = _take_device_page(session, agg)
    top = ops.topn(session, [ops.TG_BIGINT, ops.TG_DOUBLE], [1, 0], [1, 0],
                   limit)
    top.add_input(groups)
    pages = top.drain()
    elapsed = time.time() - t0
    for op in (fo, b, fl, j, agg, top):
        op.close()
    bridge.close()
    session.tpch_lineitem_free(li)
    _free_bufs(session, bufs)
    out = pages[0]
    ck = np.asarray(out[0]["values"]).astype(np.int64)
    rev = np.asarray(out[1]["values"])
    st = _strings(_lib.tg_tpch_customer_strings, sf, ck,
                  ("name", "address", "phone", "comment", "acctbal",
                   "nationkey"))
    rows = []
    for i in range(len(ck)):
        rows.append(dict(custkey=int(ck[i]), name=st["name"][i],
                         revenue=float(rev[i]),
                         acctbal_cents=int(st["acctbal"][i]),
                         n_name=NATIONS[st["nationkey"][i]],
                         address=st["address"][i], phone=st["phone"][i],
                         comment=st["comment"][i]))
    return dict(rows=rows, elapsed=elapsed)


def q20_gpu(session, sf, order_count=None, part_count=None):
    """TPC-H Q20 (potential part promotion): 'forest%' parts; availqty >
    0.5 * 1994-shipped qty per (part,supplier) — the correlated sum is NULL
    with no shipments, so the comparison joins INNER; CANADA suppliers,
    ordered by s_name."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    if part_count is None:
        part_count = int(200_000 * sf)
    supp_count = int(10_000 * sf)
    p_pk = _device_buffer(session, part_count * 8)
    p_nm = _device_buffer(session, part_count * 5)
    _check_lib(_lib.tg_tpch_gen_part2(session._h, sf, 1, part_count,
                                      p_pk, None, None, None, None, p_nm, None))
    n = part_count * 4
    ps_pk = _device_buffer(session, n * 8)
    ps_sk = _device_buffer(session, n * 8)
    ps_aq = _device_buffer(session, n * 4)
    _check_lib(_lib.tg_tpch_gen_partsupp(session._h, sf, 1, part_count,
                                         ps_pk, ps_sk, ps_aq, None))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier2(session._h, sf, 1, supp_count,
                                          s_sk, s_nk, None))
    li = session.tpch_lineitem(sf, 1, order_count, with_partkey=True,
                               with_suppkey=True)
    t0 = time.time()
    d_ff = _device_buffer(session, part_count)
    _check_lib(_lib.tg_tpch_part_name_flag(session._h, p_nm, part_count,
                                           COLOR_FOREST, 1, d_ff))
    ppage = ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                            (d_ff.value, ops.TG_TINYINT)],
                                           part_count))
    fp = ops.filter_project(session, ops.expr(("col", 1), ("i64", 1), "eq"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fp.add_input(ppage)
    fp.finish()
    forest = _take_device_page(session, fp)
    bridge_f = ops.JoinBridge(session)
    bf = ops.set_builder(session, bridge_f, [ops.TG_BIGINT], 0)
    bf.add_input(forest)
    bf.drain()
    # lineitem: 1994 window + forest part, grouped qty per (pk, sk)
    lpage = ops.page_from_device(session, ([(li.partkey, ops.TG_BIGINT),
                                            (li.suppkey, ops.TG_BIGINT),
                                            (li.quantity, ops.TG_DOUBLE),
                                            (li.shipdate, ops.TG_INTEGER)],
                                           li.row_count))
    fl = ops.filter_project(session,
                            ops.expr(("col", 3), ("i64", _D(1994, 1, 1)), "ge",
                                     ("col", 3), ("i64", _D(1995, 1, 1)), "lt",
                                     "and"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2))],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE])
    fl.add_input(lpage)
    fl.finish()
    lwin = _take_device_page(session, fl)
    sjf = ops.semi_join(session, bridge_f, 0)
    sjf.add_input(lwin)
    sjf.finish()
    lmark = _take_device_page(session, sjf)
    flm = ops.filter_project(session, ops.expr(("col", 3), ("i64", 1), "eq"),
                             [ops.expr(("col", 0)), ops.expr(("col", 1)),
                              ops.expr(("col", 2))],
                             [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE])
    flm.add_input(lmark)
    flm.finish()
    lsel = _take_device_page(session, flm)
    agg = ops.hash_aggregation(session, [0, 1], [ops.TG_BIGINT, ops.TG_BIGINT],
                               [(ops.AGG_SUM_F64, 2)])
    agg.add_input(lsel)
    agg.finish()
    sums = _take_device_page(session, agg)         # (pk, sk, sumq)
    bridge_q = ops.JoinBridge(session)
    bq = ops.hash_builder(session, bridge_q,
                          [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE],
                          [0, 1], [2])
    bq.add_input(sums)
    bq.drain()
    # partsupp of forest parts joined (INNER) with the sums
    pspage = ops.page_from_device(session, ([(ps_pk.value, ops.TG_BIGINT),
                                             (ps_sk.value, ops.TG_BIGINT),
                                             (ps_aq.value, ops.TG_INTEGER)], n))
    sjp = ops.semi_join(session, bridge_f, 0)
    sjp.add_input(pspage)
    sjp.finish()
    pmark = _take_device_page(session, sjp)
    fpm = ops.filter_project(session, ops.expr(("col", 3), ("i64", 1), "eq"),
                             [ops.expr(("col", 0)), ops.expr(("col", 1)),
                              ops.expr(("col", 2))],
                             [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_INTEGER])
    fpm.add_input(pmark)
    fpm.finish()
    ps_f = _take_device_page(session, fpm)
    jq = ops.lookup_join(session, bridge_q,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_INTEGER],
                         [0, 1], [1, 2])           # emit (sk, aq) + sumq
    jq.add_input(ps_f)
    jq.finish()
    wsum = _take_device_page(session, jq)          # (sk, aq, sumq)
    fcmp = ops.filter_project(session,
                              ops.expr(("col", 1), ("f64", 0.5), ("col", 2),
                                       "mul", "gt"),
                              [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fcmp.add_input(wsum)
    fcmp.finish()
    goodsk = _take_device_page(session, fcmp)
    dd = ops.hash_aggregation(session, [0], [ops.TG_BIGINT], [])
    dd.add_input(goodsk)
    pages_sk = dd.drain()
    elapsed = time.time() - t0
    for op in (fp, bf, fl, sjf, flm, agg, bq, sjp, fpm, jq, fcmp, dd):
        op.close()
    bridge_f.close()
    bridge_q.close()
    session.tpch_lineitem_free(li)
    for p in (p_pk, p_nm, ps_pk, ps_sk, ps_aq, d_ff):
        _device_free(session, p)
    # CANADA suppliers among the distinct keys (supplier table host-side:
    # nationkey via strings helper), sorted by name == suppkey
    sk_all = np.asarray(pages_sk[0][0]["values"]).astype(np.int64)
    st = _strings(_lib.tg_tpch_supplier_strings, sf, sk_all,
                  ("nationkey",)) if len(sk_all) else {"nationkey": np.empty(0, np.int32)}
    keep = sorted(int(k) for k, nk in zip(sk_all, st["nationkey"]) if nk == 3)
    out = _strings(_lib.tg_tpch_supplier_strings, sf, keep,
                   ("name", "address"))
    for p in (s_sk, s_nk):
        _device_free(session, p)
    return dict(names=out["name"], addresses=out["address"],
                suppkeys=keep, elapsed=elapsed)


def q9_gpu(session, sf, order_count=None, part_count=None):
    """TPC-H Q9 (product type profit): '%green%' parts; profit =
    l_extendedprice*(1-l_discount) - ps_supplycost*l_quantity grouped by
    (supplier nation, order year). Composite (pk,sk) join against partsupp;
    year from o_orderdate by epoch-day bins."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    if part_count is None:
        part_count = int(200_000 * sf)
    supp_count = int(10_000 * sf)
    p_pk = _device_buffer(session, part_count * 8)
    p_nm = _device_buffer(session, part_count * 5)
    _check_lib(_lib.tg_tpch_gen_part2(session._h, sf, 1, part_count,
                                      p_pk, None, None, None, None, p_nm, None))
    n = part_count * 4
    ps_pk = _device_buffer(session, n * 8)
    ps_sk = _device_buffer(session, n * 8)
    ps_sc = _device_buffer(session, n * 8)
    _check_lib(_lib.tg_tpch_gen_partsupp(session._h, sf, 1, part_count,
                                         ps_pk, ps_sk, None, ps_sc))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier2(session._h, sf, 1, supp_count,
                                          s_sk, s_nk, None))
    obufs = _gen_orders3(session, sf, order_count, ("ok", "od"))
    li = session.tpch_lineitem(sf, 1, order_count, with_orderkey=True,
                               with_partkey=True, with_suppkey=True)
    t0 = time.time()
    d_gf = _device_buffer(session, part_count)
    _check_lib(_lib.tg_tpch_part_name_flag(session._h, p_nm, part_count,
                                           COLOR_GREEN, 0, d_gf))
    ppage = ops.page_from_device(session, ([(p_pk.value, ops.TG_BIGINT),
                                            (d_gf.value, ops.TG_TINYINT)],
                                           part_count))
    fp = ops.filter_project(session, ops.expr(("col", 1), ("i64", 1), "eq"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fp.add_input(ppage)
    fp.finish()
    green = _take_device_page(session, fp)
    bridge_g = ops.JoinBridge(session)
    bg = ops.set_builder(session, bridge_g, [ops.TG_BIGINT], 0)
    bg.add_input(green)
    bg.drain()
    # partsupp (green parts): composite build (pk, sk) -> cost
    pspage = ops.page_from_device(session, ([(ps_pk.value, ops.TG_BIGINT),
                                             (ps_sk.value, ops.TG_BIGINT),
                                             (ps_sc.value, ops.TG_BIGINT)], n))
    fps = ops.filter_project_df(session, None,
                                [ops.expr(("col", 0)), ops.expr(("col", 1)),
                                 ops.expr(("col", 2))],
                                [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT],
                                bridge_g, 0)
    fps.add_input(pspage)
    fps.finish()
    ps_green = _take_device_page(session, fps)
    bridge_ps = ops.JoinBridge(session)
    bps = ops.hash_builder(session, bridge_ps,
                           [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT],
                           [0, 1], [2])
    bps.add_input(ps_green)
    bps.drain()
    # supplier: (sk -> nationkey) build
    sn = ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                         (s_nk.value, ops.TG_TINYINT)],
                                        supp_count))
    bridge_n = ops.JoinBridge(session)
    bn = ops.hash_builder(session, bridge_n, [ops.TG_BIGINT, ops.TG_TINYINT],
                          [0], [1])
    bn.add_input(sn)
    bn.drain()
    # lineitem: green filter, join partsupp (pk,sk), orders, supplier
    lpage = ops.page_from_device(session, ([(li.partkey, ops.TG_BIGINT),
                                            (li.suppkey, ops.TG_BIGINT),
                                            (li.orderkey, ops.TG_BIGINT),
                                            (li.quantity, ops.TG_DOUBLE),
                                            (li.extendedprice, ops.TG_DOUBLE),
                                            (li.discount, ops.TG_DOUBLE)],
                                           li.row_count))
    # green-part membership fused into the scan (DF-only filter over the
    # set-builder bitmap): one pass instead of semi-join + compacting
    # filter, which each re-streamed the 600M-row page
    fl = ops.filter_project_df(session, None,
                               [ops.expr(("col", 0)), ops.expr(("col", 1)),
                                ops.expr(("col", 2)), ops.expr(("col", 3)),
                                ops.expr(("col", 4), ("f64", 1.0), ("col", 5),
                                         "sub", "mul")],
                               [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT,
                                ops.TG_DOUBLE, ops.TG_DOUBLE],
                               bridge_g, 0)
    fl.add_input(lpage)
    fl.finish()
    lg = _take_device_page(session, fl)     # (pk, sk, ok, qty, rev)
    jps = ops.lookup_join(session, bridge_ps,
                          [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT,
                           ops.TG_DOUBLE, ops.TG_DOUBLE], [0, 1],
                          [1, 2, 3, 4])
    jps.add_input(lg)
    jps.finish()
    wcost = _take_device_page(session, jps)  # (sk, ok, qty, rev, cost_cents)
    # join orders for the year: BUILD the (smaller) green-lineitem side and
    # PROBE orders, with a fused dynamic filter pruning non-green orders
    # before the year projection (was: build 150M orders rows)
    bridge_w = ops.JoinBridge(session)
    ops.request_bitmap(bridge_w)
    bw = ops.hash_builder(session, bridge_w,
                          [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE,
                           ops.TG_DOUBLE, ops.TG_BIGINT], [1], [0, 2, 3, 4])
    bw.add_input(wcost)
    bw.drain()
    opage = ops.page_from_device(session, ([(obufs["ok"].value, ops.TG_BIGINT),
                                            (obufs["od"].value, ops.TG_INTEGER)],
                                           order_count))
    jan = [_D(y, 1, 1) for y in range(1993, 2000)]
    yexpr = [("col", 1), ("i64", jan[0]), "ge"]
    for j_ in jan[1:]:
        yexpr += [("col", 1), ("i64", j_), "ge", "add"]
    fo = ops.filter_project_df(session, None,
                               [ops.expr(("col", 0)), ops.expr(*yexpr)],
                               [ops.TG_BIGINT, ops.TG_BIGINT], bridge_w, 0)
    fo.add_input(opage)
    fo.finish()
    oyear = _take_device_page(session, fo)    # (ok, yearbin) green orders only
    jo = ops.lookup_join(session, bridge_w,
                         [ops.TG_BIGINT, ops.TG_BIGINT], [0],
                         [1])                  # emit yearbin + (sk,qty,rev,cost)
    jo.add_input(oyear)
    jo.finish()
    wyear = _take_device_page(session, jo)   # (yearbin, sk, qty, rev, cost)
    jn = ops.lookup_join(session, bridge_n,
                         [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE,
                          ops.TG_DOUBLE, ops.TG_BIGINT], [1],
                         [0, 2, 3, 4])
    jn.add_input(wyear)
    jn.finish()
    wn = _take_device_page(session, jn)      # (yearbin, qty, rev, cost, nk)
    # amount = rev - (cost/100.0) * qty  (same IEEE ops as the reference:
    # supplycost DOUBLE times quantity DOUBLE)
    # packed (nation*8 + yearbin) key + LDS-privatized dense exact sums:
    # the 175-group hash aggregation's hot-group atomics cost 17.5 ms +
    # 4.3 ms group-id assignment over ~30M rows (profiles/profq9)
    fa = ops.filter_project(session, None,
                            [ops.expr(("col", 4), ("i64", 8), "mul",
                                      ("col", 0), "add"),
                             ops.expr(("col", 2),
                                      ("col", 3), ("f64", 100.0), "div",
                                      ("col", 1), "mul", "sub")],
                            [ops.TG_BIGINT, ops.TG_DOUBLE])
    fa.add_input(wn)
    fa.finish()
    amounts = _take_device_page(session, fa)   # (nk*8+yearbin, amount)
    agg = ops.dense_aggregation(session, 0, 0, 25 * 8 - 1,
                                (ops.AGG_SUM_F64_EXACT, 1, 40))
    agg.add_input(amounts)
    pages = agg.drain()
    elapsed = time.time() - t0
    for op in (fp, bg, fps, bps, bw, fo, bn, fl, jps, jo, jn, fa, agg):
        op.close()
    for br in (bridge_g, bridge_ps, bridge_w, bridge_n):
        br.close()
    session.tpch_lineitem_free(li)
    _free_bufs(session, obufs)
    for p in (p_pk, p_nm, ps_pk, ps_sk, ps_sc, s_sk, s_nk, d_gf):
        _device_free(session, p)
    out = pages[0]
    packed = np.asarray(out[0]["values"]).astype(np.int64)
    nk = packed // 8
    yb = packed % 8
    amt = np.asarray(out[1]["values"])
    rows = sorted(((NATIONS[nk[i]], int(1992 + yb[i]), float(amt[i]))
                   for i in range(len(packed))), key=lambda r: (r[0], -r[1]))
    return dict(rows=rows, elapsed=elapsed)


def q21_gpu(session, sf, order_count=None, limit=100):
    """TPC-H Q21 (suppliers who kept orders waiting): per F-status order,
    the EXISTS/NOT EXISTS pair reduces to min/max supplier aggregates
    (>=2 distinct suppliers overall <=> minAll != maxAll; exactly one
    distinct late supplier <=> minLate == maxLate); numwait counts late
    LINES of qualifying orders per SAUDI ARABIA supplier. Lineitem is
    orderkey-clustered, so both aggregations stream (run-segmented)."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    supp_count = int(10_000 * sf)
    obufs = _gen_orders3(session, sf, order_count, ("ok", "status"))
    s_sk = _device_buffer(session, supp_count * 8)
    s_nk = _device_buffer(session, supp_count)
    _check_lib(_lib.tg_tpch_gen_supplier2(session._h, sf, 1, supp_count,
                                          s_sk, s_nk, None))
    li = session.tpch_lineitem(sf, 1, order_count, with_orderkey=True,
                               with_suppkey=True, with_dates=True)
    t0 = time.time()
    lpage = ops.page_from_device(session, ([(li.orderkey, ops.TG_BIGINT),
                                            (li.suppkey, ops.TG_BIGINT),
                                            (li.commitdate, ops.TG_INTEGER),
                                            (li.receiptdate, ops.TG_INTEGER)],
                                           li.row_count))
    # ONE streaming pass (orderkey-clustered): unmasked min/max supplier over
    # all lines + MASKED (receiptdate > commitdate) min/max/count over late
    # lines — the filter clause lowered into the accumulators. Replaces the
    # round-1 two-aggregation + 150M-row join (114 ms) and the interpreter-
    # projection variant (245 ms).
    a_all = ops.streaming_aggregation(
        session, 0,
        [(ops.AGG_MIN_I64, 1), (ops.AGG_MAX_I64, 1),
         (ops.AGG_MIN_I64, 1, 0, 3, 2), (ops.AGG_MAX_I64, 1, 0, 3, 2),
         (ops.AGG_COUNT_STAR, -1, 0, 3, 2)])
    a_all.add_input(lpage)
    a_all.finish()
    allmm = _take_device_page(session, a_all)  # (ok,minA,maxA,minL,maxL,nlate)
    # orders with late lines only; reorder to the (ok,minL,maxL,nlate,minA,
    # maxA) layout the qualify filter expects. Orders with no late lines
    # carry min/max identities in minL/maxL and 0 in nlate.
    fl = ops.filter_project(session, ops.expr(("col", 5), ("i64", 0), "gt"),
                            [ops.expr(("col", 0)), ops.expr(("col", 3)),
                             ops.expr(("col", 4)), ops.expr(("col", 5)),
                             ops.expr(("col", 1)), ops.expr(("col", 2))],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT,
                             ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT])
    fl.add_input(allmm)
    fl.finish()
    both = _take_device_page(session, fl)  # (ok, minL, maxL, nlate, minA, maxA)
    # F-status orders
    opage = ops.page_from_device(session, ([(obufs["ok"].value, ops.TG_BIGINT),
                                            (obufs["status"].value, ops.TG_TINYINT)],
                                           order_count))
    fo = ops.filter_project(session, ops.expr(("col", 1), ("i64", 0), "eq"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fo.add_input(opage)
    fo.finish()
    fords = _take_device_page(session, fo)
    bridge_o = ops.JoinBridge(session)
    bo = ops.set_builder(session, bridge_o, [ops.TG_BIGINT], 0)
    bo.add_input(fords)
    bo.drain()
    sj = ops.semi_join(session, bridge_o, 0)
    sj.add_input(both)
    sj.finish()
    marked = _take_device_page(session, sj)
    # qualify: F-status AND minA != maxA AND minL == maxL
    fq = ops.filter_project(session,
                            ops.expr(("col", 6), ("i64", 1), "eq",
                                     ("col", 4), ("col", 5), "ne", "and",
                                     ("col", 1), ("col", 2), "eq", "and"),
                            [ops.expr(("col", 1)), ops.expr(("col", 3))],
                            [ops.TG_BIGINT, ops.TG_BIGINT])
    fq.add_input(marked)
    fq.finish()
    qual = _take_device_page(session, fq)           # (suppkey, nlate)
    # numwait per supplier: suppkeys are dense 1..10k*SF, so direct-array
    # aggregation (one atomic per row) replaces the ~1M-group hash insert
    # (k_gt_assign measured 7.4 ms at SF100, profiles/profq21)
    agg = ops.dense_aggregation(session, 0, 1, supp_count,
                                (ops.AGG_SUM_I64, 1))
    agg.add_input(qual)
    agg.finish()
    counts = _take_device_page(session, agg)        # (sk, numwait)
    # SAUDI ARABIA suppliers (nation 20)
    spage = ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                            (s_nk.value, ops.TG_TINYINT)],
                                           supp_count))
    fs = ops.filter_project(session, ops.expr(("col", 1), ("i64", 20), "eq"),
                            [ops.expr(("col", 0))], [ops.TG_BIGINT])
    fs.add_input(spage)
    fs.finish()
    saudi = _take_device_page(session, fs)
    bridge_s = ops.JoinBridge(session)
    bs = ops.set_builder(session, bridge_s, [ops.TG_BIGINT], 0)
    bs.add_input(saudi)
    bs.drain()
    sj2 = ops.semi_join(session, bridge_s, 0)
    sj2.add_input(counts)
    sj2.finish()
    cm = _take_device_page(session, sj2)
    ff = ops.filter_project(session, ops.expr(("col", 2), ("i64", 1), "eq"),
                            [ops.expr(("col", 0)), ops.expr(("col", 1))],
                            [ops.TG_BIGINT, ops.TG_BIGINT])
    ff.add_input(cm)
    ff.finish()
    sc = _take_device_page(session, ff)
    # ORDER BY numwait DESC, s_name (= suppkey) ASC LIMIT 100
    top = ops.topn(session, [ops.TG_BIGINT, ops.TG_BIGINT], [1, 0], [1, 0],
                   limit)
    top.add_input(sc)
    pages = top.drain()
    elapsed = time.time() - t0
    for op in (a_all, fl, fo, bo, sj, fq, agg, fs, bs, sj2, ff, top):
        op.close()
    for br in (bridge_o, bridge_s):
        br.close()
    session.tpch_lineitem_free(li)
    _free_bufs(session, obufs)
    for p in (s_sk, s_nk):
        _device_free(session, p)
    out = pages[0]
    sk = np.asarray(out[0]["values"]).astype(np.int64)
    nw = np.asarray(out[1]["values"]).astype(np.int64)
    names = [f"Supplier#{k:09d}" for k in sk]
    return dict(names=names, numwait=nw, suppkeys=sk, elapsed=elapsed)
