"""Reproduce the in-sweep q15 slowdown (q15 measures ~17.8 ms standalone,
~85 ms inside the 22-query sweep): run the sweep-order predecessors in the
same session, then q15 three times with per-stage timing."""
import sys
import time

sys.path.insert(0, ".")
import numpy as np

import trino_amd
from trino_amd import ops
from trino_amd import tpch_queries as q
from trino_amd.tpch_queries import DATE_1996_01_01, DATE_1996_04_01, _take_device_page


def timed_q15(s, sf):
    li = s.tpch_lineitem(sf, 1, int(1_500_000 * sf), with_suppkey=True)
    t0 = time.time()
    lpage = ops.page_from_device(s, ([(li.suppkey, ops.TG_BIGINT),
                                      (li.shipdate, ops.TG_INTEGER),
                                      (li.extendedprice, ops.TG_DOUBLE),
                                      (li.discount, ops.TG_DOUBLE)],
                                     li.row_count))
    f = ops.filter_project(s,
                           ops.expr(("col", 1), ("i64", DATE_1996_01_01), "ge",
                                    ("col", 1), ("i64", DATE_1996_04_01), "lt", "and"),
                           [ops.expr(("col", 0)),
                            ops.expr(("col", 2), ("f64", 1.0), ("col", 3), "sub", "mul")],
                           [ops.TG_BIGINT, ops.TG_DOUBLE])
    f.add_input(lpage)
    f.finish()
    sel = _take_device_page(s, f)
    t1 = time.time()
    agg = ops.hash_aggregation(s, [0], [ops.TG_BIGINT],
                               [(ops.AGG_SUM_F64_EXACT, 1, 43)])
    agg.add_input(sel)
    pages = agg.drain()
    t2 = time.time()
    f.close()
    agg.close()
    s.tpch_lineitem_free(li)
    n_groups = len(pages[0][0]["values"])
    return (t1 - t0) * 1e3, (t2 - t1) * 1e3, sel.position_count, n_groups


def main(sf=100.0):
    s = trino_amd.Session(0)
    print("-- fresh session, q15 x3:")
    for i in range(3):
        fm, am, nsel, ng = timed_q15(s, sf)
        print(f"   q15: filter {fm:7.2f} ms  agg {am:7.2f} ms  "
              f"(sel={nsel} groups={ng})", flush=True)
    names = [n for n in ("q2", "q3", "q4", "q5", "q7", "q8", "q6", "q9",
                         "q10", "q11", "q12", "q13", "q14", "q16", "q17",
                         "q18", "q19", "q20", "q21", "q22")]
    for name in names:
        t0 = time.time()
        getattr(q, f"{name}_gpu")(s, sf)
        print(f"-- {name}: {(time.time()-t0)*1e3:7.1f} ms   pool {_mem(s)}",
              flush=True)
    print("-- after the full cycle, q15 x3:")
    for i in range(3):
        fm, am, nsel, ng = timed_q15(s, sf)
        print(f"   q15: filter {fm:7.2f} ms  agg {am:7.2f} ms  pool {_mem(s)}",
              flush=True)
    s.close()


def _mem(s):
    import trino_amd
    tot, cach = trino_amd.session_memory(s)
    return f"total {tot/2**30:.1f}G cached {cach/2**30:.1f}G"


if __name__ == "__main__":
    main(float(sys.argv[1]) if len(sys.argv) > 1 else 100.0)
