"""Per-stage wall timing for the Q16 pipeline (scratch tool; stages sync
internally so time.time deltas are kernel-accurate). Usage:
    python tools/profile_q16.py [sf]
"""
import sys
import time

import numpy as np

sys.path.insert(0, ".")
import ctypes

import trino_amd
from trino_amd import ops
from trino_amd.ops import _lib
from trino_amd.tpch_queries import (_check_lib, _device_buffer, _device_free,
                                    _take_device_page)


def main(sf=100.0):
    s = trino_amd.Session(0)
    part_count = int(200_000 * sf)
    supp_count = int(10_000 * sf)
    p_pk = _device_buffer(s, part_count * 8)
    p_ty = _device_buffer(s, part_count * 2)
    p_br = _device_buffer(s, part_count)
    p_sz = _device_buffer(s, part_count * 4)
    _check_lib(_lib.tg_tpch_gen_part2(s._h, sf, 1, part_count,
                                      p_pk, p_ty, p_br, p_sz, None, None, None))
    ps_pk = _device_buffer(s, part_count * 4 * 8)
    ps_sk = _device_buffer(s, part_count * 4 * 8)
    _check_lib(_lib.tg_tpch_gen_partsupp(s._h, sf, 1, part_count,
                                         ps_pk, ps_sk, None, None))
    for rep in range(3):
        marks = [("start", time.time())]

        def mark(name):
            marks.append((name, time.time()))

        d_soff = ctypes.c_void_p()
        d_sbytes = ctypes.c_void_p()
        _check_lib(_lib.tg_tpch_gen_supplier_comments(
            s._h, sf, 1, supp_count, ctypes.byref(d_soff), ctypes.byref(d_sbytes)))
        mark("gen_supp_comments")
        d_cflag = _device_buffer(s, supp_count)
        ops.varchar_like_flags(s, d_sbytes, d_soff, supp_count,
                               "%Customer%Complaints%", d_cflag)
        mark("varchar_like")
        flags = np.empty(supp_count, np.uint8)
        from trino_amd import copy_dtoh
        copy_dtoh(s, flags, d_cflag)
        bad = np.nonzero(flags)[0].astype(np.int64) + 1
        bridge_bad = ops.JoinBridge(s)
        bb = ops.set_builder(s, bridge_bad, [ops.TG_BIGINT], 0)
        bb.add_input(ops.page_from_numpy([np.ascontiguousarray(bad)]))
        bb.drain()
        mark("bad_supp_set")
        ppage = ops.page_from_device(s, ([(p_pk.value, ops.TG_BIGINT),
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
        fp = ops.filter_project(s, fexpr,
                                [ops.expr(("col", 0)), ops.expr(("col", 1)),
                                 ops.expr(("col", 2)), ops.expr(("col", 3))],
                                [ops.TG_BIGINT, ops.TG_TINYINT, ops.TG_SMALLINT,
                                 ops.TG_INTEGER])
        fp.add_input(ppage)
        fp.finish()
        part_sel = _take_device_page(s, fp)
        mark("part_filter")
        bridge_p = ops.JoinBridge(s)
        bp = ops.hash_builder(s, bridge_p, [ops.TG_BIGINT, ops.TG_TINYINT,
                                            ops.TG_SMALLINT, ops.TG_INTEGER],
                              [0], [1, 2, 3])
        bp.add_input(part_sel)
        bp.drain()
        mark("part_build")
        pspage = ops.page_from_device(s, ([(ps_pk.value, ops.TG_BIGINT),
                                           (ps_sk.value, ops.TG_BIGINT)],
                                          part_count * 4))
        sj = ops.semi_join(s, bridge_bad, 1)
        sj.add_input(pspage)
        sj.finish()
        marked = _take_device_page(s, sj)
        mark("semi_join_badsupp")
        fnb = ops.filter_project(s, ops.expr(("col", 2), ("i64", 0), "eq"),
                                 [ops.expr(("col", 0)), ops.expr(("col", 1))],
                                 [ops.TG_BIGINT, ops.TG_BIGINT])
        fnb.add_input(marked)
        fnb.finish()
        ps_ok = _take_device_page(s, fnb)
        mark("filter_badsupp")
        j = ops.lookup_join(s, bridge_p, [ops.TG_BIGINT, ops.TG_BIGINT],
                            [0], [1])
        j.add_input(ps_ok)
        j.finish()
        joined = _take_device_page(s, j)
        mark(f"lookup_join (rows={joined.position_count})")
        fpk = ops.filter_project(s, None,
                                 [ops.expr(("col", 1), ("i64", 160), "mul",
                                           ("col", 2), "add", ("i64", 64), "mul",
                                           ("col", 3), "add",
                                           ("i64", 1 << 32), "mul",
                                           ("col", 0), "add")],
                                 [ops.TG_BIGINT])
        fpk.add_input(joined)
        fpk.finish()
        packed = _take_device_page(s, fpk)
        mark("pack_key")
        n_packed = packed.position_count
        d_dedup = _device_buffer(s, max(n_packed, 1) * 8)
        n_uniq = ops.dedup_i64(s, packed.blocks[0].data, n_packed,
                               d_dedup, bits=52)
        dedup = ops.page_from_device(s, ([(d_dedup.value, ops.TG_BIGINT)],
                                         n_uniq))
        mark(f"dedup_sort (groups={n_uniq})")
        fg = ops.filter_project(s, None,
                                [ops.expr(("col", 0), ("i64", 1 << 32), "div")],
                                [ops.TG_BIGINT])
        fg.add_input(dedup)
        fg.finish()
        combos = _take_device_page(s, fg)
        mark("unpack_combo")
        d2 = ops.hash_aggregation(s, [0], [ops.TG_BIGINT],
                                  [(ops.AGG_COUNT_STAR, -1)])
        d2.add_input(combos)
        pages = d2.drain()
        mark(f"count_per_combo (combos={len(pages[0][0]['values'])})")

        total = (marks[-1][1] - marks[0][1]) * 1e3
        print(f"--- rep {rep}: total {total:.1f} ms")
        for (n0, t0), (n1, t1) in zip(marks, marks[1:]):
            print(f"  {n1:<44s} {(t1 - t0) * 1e3:8.2f} ms")
        for op in (bb, fp, bp, sj, fnb, j, fpk, fg, d2):
            op.close()
        bridge_bad.close()
        bridge_p.close()
        _device_free(s, d_cflag)
        _device_free(s, d_dedup)
    for p in (p_pk, p_ty, p_br, p_sz, ps_pk, ps_sk):
        _device_free(s, p)
    s.close()


if __name__ == "__main__":
    main(float(sys.argv[1]) if len(sys.argv) > 1 else 100.0)
