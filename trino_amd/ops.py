"""trino_amd.ops — ctypes bindings for the operator-level C-ABI
(include/trino_gpu.h: Operator/Page/Block/SelectedPositions surfaces).
Product plumbing: builds descriptors from numpy arrays, drives the
addInput/getOutput state machine, downloads device output pages.
"""
import ctypes

import numpy as np

from . import _lib, _check, Session

TG_BIGINT, TG_INTEGER, TG_SMALLINT, TG_TINYINT, TG_DOUBLE, TG_DATE, TG_BOOLEAN, TG_VARCHAR = range(8)
STEP_PARTIAL, STEP_FINAL, STEP_SINGLE = 0, 1, 2
(AGG_COUNT_STAR, AGG_COUNT_COL, AGG_SUM_F64, AGG_SUM_I64, AGG_AVG_F64,
 AGG_SUM_F64_EXACT, AGG_MIN_I64, AGG_MAX_I64) = range(8)

_NP2TG = {np.dtype(np.int64): TG_BIGINT, np.dtype(np.int32): TG_INTEGER,
          np.dtype(np.int16): TG_SMALLINT, np.dtype(np.int8): TG_TINYINT,
          np.dtype(np.uint8): TG_TINYINT, np.dtype(np.float64): TG_DOUBLE}
_TG2NP = {TG_BIGINT: np.int64, TG_INTEGER: np.int32, TG_SMALLINT: np.int16,
          TG_TINYINT: np.int8, TG_DOUBLE: np.float64, TG_DATE: np.int32,
          TG_BOOLEAN: np.int8}


class TgBlock(ctypes.Structure):
    pass


TgBlock._fields_ = [
    ("type", ctypes.c_int32), ("kind", ctypes.c_int32),
    ("position_count", ctypes.c_int64),
    ("on_device", ctypes.c_int32), ("_pad", ctypes.c_int32),
    ("data", ctypes.c_void_p), ("valid", ctypes.c_void_p),
    ("offsets", ctypes.c_void_p), ("ids", ctypes.c_void_p),
    ("dictionary", ctypes.POINTER(TgBlock)),
]


class TgPage(ctypes.Structure):
    _fields_ = [("channel_count", ctypes.c_int32),
                ("position_count", ctypes.c_int64),
                ("blocks", ctypes.POINTER(TgBlock))]


class TgSelected(ctypes.Structure):
    _fields_ = [("is_list", ctypes.c_int32), ("offset", ctypes.c_int32),
                ("size", ctypes.c_int32), ("positions", ctypes.c_void_p)]


class TgExprInst(ctypes.Structure):
    class _Imm(ctypes.Union):
        _fields_ = [("f64", ctypes.c_double), ("i64", ctypes.c_int64)]
    _fields_ = [("op", ctypes.c_int32), ("arg0", ctypes.c_int32), ("imm", _Imm)]


class TgExpr(ctypes.Structure):
    _fields_ = [("insts", ctypes.POINTER(TgExprInst)), ("count", ctypes.c_int32)]


class TgAggSpec(ctypes.Structure):
    _fields_ = [("fn", ctypes.c_int32), ("input_channel", ctypes.c_int32),
                ("scale_pow", ctypes.c_int32),
                ("mask_gt_a", ctypes.c_int32), ("mask_gt_b", ctypes.c_int32),
                ("_pad", ctypes.c_int32)]


(OP_COL, OP_CONST_F64, OP_CONST_I64, OP_ADD, OP_SUB, OP_MUL, OP_DIV,
 OP_LE, OP_LT, OP_GE, OP_GT, OP_EQ, OP_NE, OP_AND, OP_OR, OP_NOT,
 OP_BETWEEN, OP_IN) = range(18)


def expr(*postfix):
    """Build a TgExpr from postfix tuples: ('col', i) ('f64', v) ('i64', v)
    or op names 'le','mul','and',..."""
    ops = {"add": OP_ADD, "sub": OP_SUB, "mul": OP_MUL, "div": OP_DIV,
           "le": OP_LE, "lt": OP_LT, "ge": OP_GE, "gt": OP_GT,
           "eq": OP_EQ, "ne": OP_NE, "and": OP_AND, "or": OP_OR,
           "not": OP_NOT, "between": OP_BETWEEN}
    insts = (TgExprInst * len(postfix))()
    for i, it in enumerate(postfix):
        if isinstance(it, tuple):
            kind, v = it
            if kind == "col":
                insts[i].op = OP_COL
                insts[i].arg0 = v
            elif kind == "f64":
                insts[i].op = OP_CONST_F64
                insts[i].imm.f64 = float(v)
            elif kind == "i64":
                insts[i].op = OP_CONST_I64
                insts[i].imm.i64 = int(v)
            else:
                raise ValueError(kind)
        else:
            insts[i].op = ops[it]
    e = TgExpr()
    e.insts = insts
    e.count = len(postfix)
    e._keepalive = insts
    return e


def page_from_numpy(columns, valids=None):
    """columns: list of 1-D numpy arrays (host). valids: optional list of
    packed uint64 bitmaps (bit=1 valid) or None per column."""
    n = len(columns[0])
    blocks = (TgBlock * len(columns))()
    keep = []
    for i, col in enumerate(columns):
        assert col.flags["C_CONTIGUOUS"]
        blocks[i].type = _NP2TG[col.dtype]
        blocks[i].kind = 0
        blocks[i].position_count = n
        blocks[i].on_device = 0
        blocks[i].data = col.ctypes.data
        v = valids[i] if valids else None
        blocks[i].valid = v.ctypes.data if v is not None else None
        keep.append((col, v))
    p = TgPage()
    p.channel_count = len(columns)
    p.position_count = n
    p.blocks = blocks
    p._keepalive = (blocks, keep)
    return p


_lib.tg_operator_needs_input.restype = ctypes.c_int
_lib.tg_operator_needs_input.argtypes = [ctypes.c_void_p]
_lib.tg_operator_add_input.restype = ctypes.c_int
_lib.tg_operator_add_input.argtypes = [ctypes.c_void_p, ctypes.POINTER(TgPage)]
_lib.tg_operator_get_output.restype = ctypes.c_int
_lib.tg_operator_get_output.argtypes = [ctypes.c_void_p, ctypes.POINTER(TgPage),
                                        ctypes.POINTER(ctypes.c_int)]
_lib.tg_operator_finish.restype = ctypes.c_int
_lib.tg_operator_finish.argtypes = [ctypes.c_void_p]
_lib.tg_operator_close.restype = None
_lib.tg_operator_close.argtypes = [ctypes.c_void_p]
_lib.tg_filter_project_create.restype = ctypes.c_int
_lib.tg_filter_project_create.argtypes = [ctypes.c_void_p, ctypes.POINTER(TgExpr),
                                          ctypes.POINTER(TgExpr), ctypes.c_void_p,
                                          ctypes.c_int32, ctypes.c_void_p]
_lib.tg_filter_run.restype = ctypes.c_int
_lib.tg_filter_run.argtypes = [ctypes.c_void_p, ctypes.POINTER(TgExpr),
                               ctypes.POINTER(TgPage), ctypes.POINTER(TgSelected),
                               ctypes.c_void_p, ctypes.c_void_p]
_lib.tg_streaming_aggregation_create.restype = ctypes.c_int
_lib.tg_streaming_aggregation_create.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                                 ctypes.c_void_p, ctypes.c_int32,
                                                 ctypes.c_int32, ctypes.c_void_p]
_lib.tg_hash_aggregation_create.restype = ctypes.c_int
_lib.tg_hash_aggregation_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                            ctypes.c_int32, ctypes.c_void_p,
                                            ctypes.c_void_p, ctypes.c_int32,
                                            ctypes.c_int32, ctypes.c_void_p]
_lib.tg_join_bridge_create.restype = ctypes.c_int
_lib.tg_join_bridge_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
_lib.tg_join_bridge_close.restype = None
_lib.tg_join_bridge_close.argtypes = [ctypes.c_void_p]
_lib.tg_hash_builder_create.restype = ctypes.c_int
_lib.tg_hash_builder_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p] + [ctypes.c_void_p, ctypes.c_int32] * 3 + [ctypes.c_void_p]
_lib.tg_lookup_join_create.restype = ctypes.c_int
_lib.tg_lookup_join_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p] + [ctypes.c_void_p, ctypes.c_int32] * 3 + [ctypes.c_void_p]
_lib.tg_lookup_join_create_ex.restype = ctypes.c_int
_lib.tg_lookup_join_create_ex.argtypes = [ctypes.c_void_p, ctypes.c_void_p] + [ctypes.c_void_p, ctypes.c_int32] * 3 + [ctypes.c_int32, ctypes.c_void_p]
_lib.tg_page_partitioner_create.restype = ctypes.c_int
_lib.tg_page_partitioner_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                            ctypes.c_int32, ctypes.c_void_p,
                                            ctypes.c_int32, ctypes.c_int32, ctypes.c_void_p]
_lib.tg_page_partitioner_get_partition.restype = ctypes.c_int
_lib.tg_page_partitioner_get_partition.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                                   ctypes.POINTER(TgPage)]
_lib.tg_hash_rows.restype = ctypes.c_int
_lib.tg_hash_rows.argtypes = [ctypes.c_void_p, ctypes.POINTER(TgPage),
                              ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p]


def _i32arr(lst):
    return np.array(lst, np.int32)


class Operator:
    """operator/Operator.java:18-50 state machine over the C ABI."""

    def __init__(self, session, handle):
        self.s = session
        self._h = handle

    def add_input(self, page):
        _check(_lib.tg_operator_add_input(self._h, ctypes.byref(page)))

    def finish(self):
        _check(_lib.tg_operator_finish(self._h))

    def get_output(self):
        """Returns (columns_as_numpy_or_None, finished). Device blocks are
        downloaded to host numpy for inspection."""
        out = TgPage()
        fin = ctypes.c_int(0)
        _check(_lib.tg_operator_get_output(self._h, ctypes.byref(out), ctypes.byref(fin)))
        if out.channel_count == 0:
            return None, bool(fin.value)
        return _download_page(self.s, out), bool(fin.value)

    def drain(self):
        """finish + collect all output pages (list of column dicts)."""
        self.finish()
        pages = []
        while True:
            cols, fin = self.get_output()
            if cols is not None:
                pages.append(cols)
            else:
                if fin:
                    return pages

    def close(self):
        if self._h:
            _lib.tg_operator_close(self._h)
            self._h = None


def _download_page(session, page):
    from . import copy_dtoh
    cols = []
    for c in range(page.channel_count):
        b = page.blocks[c]
        n = b.position_count
        valid = None
        if b.valid:
            valid = np.empty(max((n + 63) // 64, 1), np.uint64)
            if n:
                copy_dtoh(session, valid, b.valid)
        if b.type == TG_VARCHAR:
            offsets = np.empty(n + 1, np.int32)
            copy_dtoh(session, offsets, b.offsets)
            nbytes = int(offsets[n])
            data = np.empty(max(nbytes, 1), np.uint8)
            if nbytes:
                copy_dtoh(session, data, b.data)
            vals = [bytes(data[offsets[i]:offsets[i + 1]]) for i in range(n)]
            cols.append({"values": vals, "valid": valid, "type": b.type,
                         "offsets": offsets, "bytes": data[:nbytes]})
            continue
        arr = np.empty(n, _TG2NP[b.type])
        if n:
            copy_dtoh(session, arr, b.data)
        cols.append({"values": arr, "valid": valid, "type": b.type})
    return cols


def filter_project(session, filter_expr, projections, out_types=None):
    h = ctypes.c_void_p()
    n = len(projections)
    proj_arr = (TgExpr * n)()
    for i, p in enumerate(projections):
        proj_arr[i] = p
    ot = _i32arr(out_types or [TG_DOUBLE] * n)
    _check(_lib.tg_filter_project_create(
        session._h, ctypes.byref(filter_expr) if filter_expr else None,
        proj_arr, ot.ctypes.data, n, ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (proj_arr, ot, filter_expr, projections)
    return op


def filter_run(session, filter_expr, page, input_sel=None):
    cap = int(page.position_count if input_sel is None else input_sel.size)
    out = np.empty(max(cap, 1), np.int32)
    cnt = ctypes.c_int32(0)
    _check(_lib.tg_filter_run(session._h, ctypes.byref(filter_expr), ctypes.byref(page),
                              ctypes.byref(input_sel) if input_sel else None,
                              out.ctypes.data, ctypes.byref(cnt)))
    return out[:cnt.value]


def streaming_aggregation(session, key_channel, aggs, step=STEP_SINGLE):
    """StreamingAggregationOperator analog: input clustered by a single
    non-null BIGINT key. aggs: list of (fn, input_channel[, scale_pow])."""
    h = ctypes.c_void_p()
    sp = (TgAggSpec * len(aggs))()
    for i, a in enumerate(aggs):
        sp[i].fn = a[0]
        sp[i].input_channel = a[1]
        sp[i].scale_pow = a[2] if len(a) > 2 else 0
        sp[i].mask_gt_a = a[3] if len(a) > 4 else -1
        sp[i].mask_gt_b = a[4] if len(a) > 4 else -1
    _check(_lib.tg_streaming_aggregation_create(session._h, key_channel, sp,
                                                len(aggs), step, ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (sp,)
    return op


def hash_aggregation(session, group_channels, group_types, aggs, step=STEP_SINGLE):
    """aggs: list of (fn, input_channel)."""
    h = ctypes.c_void_p()
    gc = _i32arr(group_channels)
    gt = _i32arr(group_types)
    sp = (TgAggSpec * len(aggs))()
    for i, a in enumerate(aggs):
        sp[i].fn = a[0]
        sp[i].input_channel = a[1]
        sp[i].scale_pow = a[2] if len(a) > 2 else 0
        sp[i].mask_gt_a = a[3] if len(a) > 4 else -1
        sp[i].mask_gt_b = a[4] if len(a) > 4 else -1
    _check(_lib.tg_hash_aggregation_create(session._h, gc.ctypes.data, len(gc),
                                           gt.ctypes.data, sp, len(aggs), step,
                                           ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (gc, gt, sp)
    return op


class JoinBridge:
    def __init__(self, session):
        self.s = session
        self._h = ctypes.c_void_p()
        _check(_lib.tg_join_bridge_create(session._h, ctypes.byref(self._h)))

    def close(self):
        if self._h:
            _lib.tg_join_bridge_close(self._h)
            self._h = None


def hash_builder(session, bridge, build_types, key_channels, output_channels):
    h = ctypes.c_void_p()
    bt = _i32arr(build_types)
    kc = _i32arr(key_channels)
    oc = _i32arr(output_channels)
    _check(_lib.tg_hash_builder_create(session._h, bridge._h, bt.ctypes.data, len(bt),
                                       kc.ctypes.data, len(kc), oc.ctypes.data, len(oc),
                                       ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (bt, kc, oc)
    return op


def lookup_join(session, bridge, probe_types, key_channels, probe_output_channels,
                join_type=0):
    """join_type: 0 inner, 1 probe-outer (LEFT)."""
    h = ctypes.c_void_p()
    pt = _i32arr(probe_types)
    kc = _i32arr(key_channels)
    oc = _i32arr(probe_output_channels)
    _check(_lib.tg_lookup_join_create_ex(session._h, bridge._h, pt.ctypes.data, len(pt),
                                         kc.ctypes.data, len(kc), oc.ctypes.data,
                                         len(oc), join_type, ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (pt, kc, oc)
    return op


def page_partitioner(session, types, partition_channels, partition_count):
    h = ctypes.c_void_p()
    ty = _i32arr(types)
    pc = _i32arr(partition_channels)
    _check(_lib.tg_page_partitioner_create(session._h, ty.ctypes.data, len(ty),
                                           pc.ctypes.data, len(pc), partition_count,
                                           ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (ty, pc)
    return op


def get_partition(session, partitioner_op, p):
    out = TgPage()
    _check(_lib.tg_page_partitioner_get_partition(partitioner_op._h, p, ctypes.byref(out)))
    if out.channel_count == 0:
        return None
    return _download_page(session, out)


def hash_rows(session, page, channels):
    out = np.empty(max(int(page.position_count), 1), np.uint64)
    ch = _i32arr(channels)
    _check(_lib.tg_hash_rows(session._h, ctypes.byref(page), ch.ctypes.data,
                             len(ch), out.ctypes.data))
    return out[:page.position_count]


def page_from_device(session, ptrs_types_n):
    """Build a device-resident page: list of (device_ptr, tg_type), n rows."""
    ptrs_types, n = ptrs_types_n
    blocks = (TgBlock * len(ptrs_types))()
    for i, (ptr, ty) in enumerate(ptrs_types):
        blocks[i].type = ty
        blocks[i].kind = 0
        blocks[i].position_count = n
        blocks[i].on_device = 1
        blocks[i].data = ptr
    p = TgPage()
    p.channel_count = len(ptrs_types)
    p.position_count = n
    p.blocks = blocks
    p._keepalive = blocks
    return p


_lib.tg_topn_create.restype = ctypes.c_int
_lib.tg_topn_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                                ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                                ctypes.c_int32, ctypes.c_void_p]
_lib.tg_join_bridge_key_range.restype = ctypes.c_int
_lib.tg_join_bridge_key_range.argtypes = [ctypes.c_void_p] + [ctypes.c_void_p] * 3


def topn(session, types, sort_channels, sort_desc, limit):
    """TopN operator (operator/TopNOperator analog): ORDER BY ... LIMIT n."""
    h = ctypes.c_void_p()
    ty = _i32arr(types)
    sc = _i32arr(sort_channels)
    sd = _i32arr(sort_desc)
    _check(_lib.tg_topn_create(session._h, ty.ctypes.data, len(ty), sc.ctypes.data,
                               sd.ctypes.data, len(sc), limit, ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (ty, sc, sd)
    return op


def join_key_range(bridge):
    """Dynamic filter source: (min, max, non_null_rows) of the build keys."""
    mn = ctypes.c_int64()
    mx = ctypes.c_int64()
    nr = ctypes.c_int64()
    _check(_lib.tg_join_bridge_key_range(bridge._h, ctypes.byref(mn),
                                         ctypes.byref(mx), ctypes.byref(nr)))
    return mn.value, mx.value, nr.value


_lib.tg_set_builder_create.restype = ctypes.c_int
_lib.tg_set_builder_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_void_p, ctypes.c_int32,
                                       ctypes.c_int32, ctypes.c_void_p]
_lib.tg_semi_join_create.restype = ctypes.c_int
_lib.tg_semi_join_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_int32, ctypes.c_void_p]


def set_builder(session, bridge, build_types, key_channel):
    """SetBuilderOperator analog: semi-join membership source (bitmap for
    dense BIGINT ranges)."""
    h = ctypes.c_void_p()
    bt = _i32arr(build_types)
    _check(_lib.tg_set_builder_create(session._h, bridge._h, bt.ctypes.data,
                                      len(bt), key_channel, ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (bt,)
    return op


def semi_join(session, bridge, key_channel):
    """HashSemiJoinOperator analog: probe page + BOOLEAN matched channel
    (NULL for null probe keys)."""
    h = ctypes.c_void_p()
    _check(_lib.tg_semi_join_create(session._h, bridge._h, key_channel, ctypes.byref(h)))
    return Operator(session, h)


def page_with_varchar(columns):
    """Build a page where entries may be numpy arrays (fixed width) or
    (bytes_array_uint8, offsets_int32) tuples for VARCHAR channels."""
    n = None
    for c in columns:
        n = len(c[1]) - 1 if isinstance(c, tuple) else len(c)
        break
    blocks = (TgBlock * len(columns))()
    keep = []
    for i, col in enumerate(columns):
        if isinstance(col, tuple):
            data, offsets = col
            blocks[i].type = TG_VARCHAR
            blocks[i].kind = 0
            blocks[i].position_count = len(offsets) - 1
            blocks[i].on_device = 0
            blocks[i].data = data.ctypes.data
            blocks[i].offsets = offsets.ctypes.data
            keep.append((data, offsets))
        else:
            blocks[i].type = _NP2TG[col.dtype]
            blocks[i].kind = 0
            blocks[i].position_count = len(col)
            blocks[i].on_device = 0
            blocks[i].data = col.ctypes.data
            keep.append(col)
    p = TgPage()
    p.channel_count = len(columns)
    p.position_count = blocks[0].position_count
    p.blocks = blocks
    p._keepalive = (blocks, keep)
    return p


# ---- round 2: extended TPC-H tables + text/LIKE (device) ----
_lib.tg_tpch_gen_part2.restype = ctypes.c_int
_lib.tg_tpch_gen_part2.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                   ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 7
_lib.tg_tpch_gen_partsupp.restype = ctypes.c_int
_lib.tg_tpch_gen_partsupp.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                      ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 4
_lib.tg_tpch_gen_supplier2.restype = ctypes.c_int
_lib.tg_tpch_gen_supplier2.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                       ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 3
_lib.tg_tpch_gen_orders3.restype = ctypes.c_int
_lib.tg_tpch_gen_orders3.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                     ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 8
_lib.tg_pool_like_flags.restype = ctypes.c_int
_lib.tg_pool_like_flags.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_void_p, ctypes.c_int64,
                                    ctypes.c_char_p, ctypes.c_void_p]
_lib.tg_varchar_like_flags.restype = ctypes.c_int
_lib.tg_varchar_like_flags.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_void_p, ctypes.c_int64,
                                       ctypes.c_char_p, ctypes.c_void_p]
_lib.tg_tpch_gen_supplier_comments.restype = ctypes.c_int
_lib.tg_tpch_gen_supplier_comments.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                               ctypes.c_int64, ctypes.c_int64,
                                               ctypes.c_void_p, ctypes.c_void_p]


def pool_like_flags(session, d_offs, d_lens, n, pattern, d_flags):
    _check(_lib.tg_pool_like_flags(session._h, d_offs, d_lens, n,
                                   pattern.encode(), d_flags))


def varchar_like_flags(session, d_bytes, d_offsets, n, pattern, d_flags):
    _check(_lib.tg_varchar_like_flags(session._h, d_bytes, d_offsets, n,
                                      pattern.encode(), d_flags))


_lib.tg_join_bridge_request_bitmap.restype = ctypes.c_int
_lib.tg_join_bridge_request_bitmap.argtypes = [ctypes.c_void_p]
_lib.tg_filter_project_create_df.restype = ctypes.c_int
_lib.tg_filter_project_create_df.argtypes = [ctypes.c_void_p,
                                             ctypes.POINTER(TgExpr),
                                             ctypes.POINTER(TgExpr),
                                             ctypes.c_void_p, ctypes.c_int32,
                                             ctypes.c_void_p, ctypes.c_int32,
                                             ctypes.c_void_p]


def filter_project_df(session, filter_expr, projections, out_types,
                      df_bridge, df_key_channel):
    """filter_project with a fused dynamic filter from a join bridge
    (request_bitmap must have been called on the bridge before build)."""
    h = ctypes.c_void_p()
    n = len(projections)
    proj_arr = (TgExpr * n)()
    for i, p in enumerate(projections):
        proj_arr[i] = p
    ot = _i32arr(out_types or [TG_DOUBLE] * n)
    _check(_lib.tg_filter_project_create_df(
        session._h, ctypes.byref(filter_expr) if filter_expr else None,
        proj_arr, ot.ctypes.data, n, df_bridge._h, df_key_channel,
        ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (proj_arr, ot, filter_expr, projections)
    return op


def request_bitmap(bridge):
    _check(_lib.tg_join_bridge_request_bitmap(bridge._h))


def page_with_dict(columns):
    """Build a page where entries may be numpy arrays (flat) or
    (dict_values_array, ids_int32_array) tuples for dictionary-encoded
    channels (DictionaryBlock analog)."""
    blocks = (TgBlock * len(columns))()
    keep = []
    n = None
    for i, col in enumerate(columns):
        if isinstance(col, tuple):
            dvals, ids = col
            n = len(ids)
            d = TgBlock()
            d.type = _NP2TG[dvals.dtype]
            d.kind = 0
            d.position_count = len(dvals)
            d.on_device = 0
            d.data = dvals.ctypes.data
            keep += [dvals, ids, d]
            blocks[i].type = d.type
            blocks[i].kind = 1            # TG_BK_DICTIONARY
            blocks[i].position_count = n
            blocks[i].on_device = 0
            blocks[i].ids = ids.ctypes.data
            blocks[i].dictionary = ctypes.pointer(d)
        else:
            n = len(col)
            blocks[i].type = _NP2TG[col.dtype]
            blocks[i].kind = 0
            blocks[i].position_count = n
            blocks[i].on_device = 0
            blocks[i].data = col.ctypes.data
            keep.append(col)
    page = TgPage()
    page.channel_count = len(columns)
    page.position_count = n
    page.blocks = ctypes.cast(blocks, ctypes.POINTER(TgBlock))
    page._keepalive = (blocks, keep)
    return page


_lib.tg_dense_aggregation_create.restype = ctypes.c_int
_lib.tg_dense_aggregation_create.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                             ctypes.c_int64, ctypes.c_int64,
                                             ctypes.c_void_p, ctypes.c_void_p]


def dense_aggregation(session, key_channel, key_min, key_max, agg):
    """Direct-array aggregation over a dense BIGINT key range; one
    (fn, input_channel[, scale_pow]) agg; output (key, value) in key order."""
    h = ctypes.c_void_p()
    sp = TgAggSpec()
    sp.fn = agg[0]
    sp.input_channel = agg[1]
    sp.scale_pow = agg[2] if len(agg) > 2 else 0
    sp.mask_gt_a = -1
    sp.mask_gt_b = -1
    _check(_lib.tg_dense_aggregation_create(session._h, key_channel, key_min,
                                            key_max, ctypes.byref(sp),
                                            ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (sp,)
    return op


_lib.tg_mark_distinct_create.restype = ctypes.c_int
_lib.tg_mark_distinct_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_int32, ctypes.c_void_p,
                                         ctypes.c_void_p]


_lib.tg_dedup_i64.restype = ctypes.c_int
_lib.tg_dedup_i64.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                              ctypes.c_int64, ctypes.c_int32,
                              ctypes.c_void_p, ctypes.c_void_p]


def dedup_i64(session, d_in, n, d_out, bits=64):
    """Sort-based DISTINCT of a device BIGINT column; returns the distinct
    count (results sorted ascending in d_out)."""
    out_n = ctypes.c_int64(0)
    _check(_lib.tg_dedup_i64(session._h, d_in, n, bits, d_out,
                             ctypes.byref(out_n)))
    return out_n.value


_lib.tg_pa_controller_create.restype = ctypes.c_int
_lib.tg_pa_controller_create.argtypes = [ctypes.c_int64, ctypes.c_double,
                                         ctypes.c_void_p]
_lib.tg_pa_controller_close.restype = None
_lib.tg_pa_controller_close.argtypes = [ctypes.c_void_p]
_lib.tg_pa_controller_disabled.restype = ctypes.c_int32
_lib.tg_pa_controller_disabled.argtypes = [ctypes.c_void_p]
_lib.tg_pa_controller_on_flush.restype = ctypes.c_int
_lib.tg_pa_controller_on_flush.argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                           ctypes.c_int64, ctypes.c_int64,
                                           ctypes.c_int32]
_lib.tg_hash_aggregation_set_controller.restype = ctypes.c_int
_lib.tg_hash_aggregation_set_controller.argtypes = [ctypes.c_void_p,
                                                    ctypes.c_void_p]


class PartialAggController:
    """PartialAggregationController analog (adaptive partial aggregation):
    shared across the PARTIAL-step hash aggregations of one plan node.
    Disables partial aggregation when unique/input rows > threshold after
    1.5x max_partial_bytes of input; re-enables after 200x more bytes."""

    def __init__(self, max_partial_bytes=16 << 20, threshold=0.8):
        h = ctypes.c_void_p()
        _check(_lib.tg_pa_controller_create(max_partial_bytes, threshold,
                                            ctypes.byref(h)))
        self._h = h

    @property
    def disabled(self):
        return bool(_lib.tg_pa_controller_disabled(self._h))

    def on_flush(self, bytes_, rows, unique_rows=0, have_unique=False):
        _check(_lib.tg_pa_controller_on_flush(self._h, bytes_, rows,
                                              unique_rows, 1 if have_unique else 0))

    def attach(self, op):
        _check(_lib.tg_hash_aggregation_set_controller(op._h, self._h))

    def close(self):
        if self._h:
            _lib.tg_pa_controller_close(self._h)
            self._h = None


def mark_distinct(session, key_channels, key_types):
    """MarkDistinctOperator analog: pass-through + BOOLEAN first-occurrence
    channel appended after the input channels."""
    h = ctypes.c_void_p()
    kc = _i32arr(key_channels)
    kt = _i32arr(key_types)
    _check(_lib.tg_mark_distinct_create(session._h, kc.ctypes.data, len(kc),
                                        kt.ctypes.data, ctypes.byref(h)))
    op = Operator(session, h)
    op._keep = (kc, kt)
    return op
