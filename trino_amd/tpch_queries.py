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
    if cust_key_exchange is None:
        cust_sel = _take_device_page(session, f1)
        b1.add_input(cust_sel)
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
    f3 = ops.filter_project(session,
                            ops.expr(("col", 1), ("i64", DATE_1995_03_15), "gt"),
                            [ops.expr(("col", 0)),
                             ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul")],
                            [ops.TG_BIGINT, ops.TG_DOUBLE])
    f3.add_input(lpage)
    f3.finish()
    li_sel = _take_device_page(session, f3)

    j2 = ops.lookup_join(session, bridge2, [ops.TG_BIGINT, ops.TG_DOUBLE],
                         [0], [0, 1])      # emit orderkey, discprice (+ build orderdate)
    j2.add_input(li_sel)
    j2.finish()
    joined = _take_device_page(session, j2)

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
                    top10=top10, elapsed=elapsed)
    orderkey = out[0]["values"]
    orderdate = out[1]["values"]
    revenue = out[2]["values"]
    return dict(orderkey=orderkey, orderdate=orderdate, revenue=revenue,
                top10=top10, elapsed=elapsed)


# ---- small device-buffer helpers over the C ABI ----
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
    agg = ops.hash_aggregation(session, [0], [ops.TG_BIGINT],
                               [(ops.AGG_SUM_F64_EXACT, 1, 43)])
    agg.add_input(sel)
    pages = agg.drain()
    elapsed = time.time() - t0
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
    # build1: customers (custkey -> nationkey)
    br1 = ops.JoinBridge(session)
    b1 = ops.hash_builder(session, br1, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b1.add_input(ops.page_from_device(session, ([(c_ck.value, ops.TG_BIGINT),
                                                 (c_nk.value, ops.TG_TINYINT)],
                                                cust_count)))
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
    b2 = ops.hash_builder(session, br2, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b2.add_input(ojoined)
    b2.drain()
    # lineitem -> (orderkey, suppkey, revenue) -> join orders
    fp = ops.filter_project(session, None,
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2), ("f64", 1.0), ("col", 3),
                                      "sub", "mul")],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE])
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
    # join supplier nation
    br3 = ops.JoinBridge(session)
    b3 = ops.hash_builder(session, br3, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b3.add_input(ops.page_from_device(session, ([(s_sk.value, ops.TG_BIGINT),
                                                 (s_nk.value, ops.TG_TINYINT)],
                                                supp_count)))
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
    for op in (b1, f1, j1, b2, fp, j2, b3, j3, f2, agg):
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
    b3 = ops.hash_builder(session, br3, [ops.TG_BIGINT, ops.TG_TINYINT], [0], [1])
    b3.add_input(ssel)
    b3.drain()
    # lineitem window -> inner join vs FR/DE suppliers
    f = ops.filter_project(session,
                           ops.expr(("col", 4), ("i64", DATE_1995_01_01), "ge",
                                    ("col", 4), ("i64", DATE_1995_01_01 + 731),
                                    "lt", "and"),
                           [ops.expr(("col", 1)), ops.expr(("col", 0)),
                            ops.expr(("col", 2), ("f64", 1.0), ("col", 3),
                                     "sub", "mul"),
                            ops.expr(("col", 4), ("i64", DATE_1996_01_01), "ge")],
                           [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_DOUBLE,
                            ops.TG_DOUBLE])
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
    # lineitem -> part filter join -> orders join -> supplier join
    fl = ops.filter_project(session, None,
                            [ops.expr(("col", 0)), ops.expr(("col", 1)),
                             ops.expr(("col", 2)),
                             ops.expr(("col", 3), ("f64", 1.0), ("col", 4),
                                      "sub", "mul")],
                            [ops.TG_BIGINT, ops.TG_BIGINT, ops.TG_BIGINT,
                             ops.TG_DOUBLE])
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
