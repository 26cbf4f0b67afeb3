"""oracle — CPU restatement of the reference's hot-path algorithms.

TEST INFRASTRUCTURE ONLY: importable by tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg. The product path (trino_amd) never imports this;
it fails loudly if its HIP extension is missing on a GPU box.

ctypes bindings over oracle/liboracle.so (built by `make -C oracle` /
__graft_entry__.build()). numpy in/out.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def _ensure_built():
    if not os.path.exists(_SO):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


_ensure_built()
_lib = ctypes.CDLL(_SO)

_lib.tpch_lineitem_count.restype = ctypes.c_int64
_lib.tpch_lineitem_count.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64]
_lib.tpch_gen_lineitem.restype = ctypes.c_int64
_lib.tpch_gen_lineitem.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 16
_lib.tpch_gen_orders.restype = ctypes.c_int64
_lib.tpch_gen_orders.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 3
_lib.tpch_gen_orders2.restype = ctypes.c_int64
_lib.tpch_gen_orders2.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 4
_lib.tpch_gen_part_cols.restype = ctypes.c_int64
_lib.tpch_gen_part_cols.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 2
_lib.tpch_gen_supplier.restype = ctypes.c_int64
_lib.tpch_gen_supplier.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 2
_lib.tpch_gen_customer.restype = ctypes.c_int64
_lib.tpch_gen_customer.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 4

_lib.o_bigint_hash.restype = ctypes.c_uint64
_lib.o_bigint_hash.argtypes = [ctypes.c_int64]
_lib.o_double_hash.restype = ctypes.c_uint64
_lib.o_double_hash.argtypes = [ctypes.c_double]
_lib.o_murmur3_mix.restype = ctypes.c_uint64
_lib.o_murmur3_mix.argtypes = [ctypes.c_uint64]
_lib.o_combine_hash.restype = ctypes.c_int64
_lib.o_combine_hash.argtypes = [ctypes.c_int64, ctypes.c_int64]
_lib.o_xxhash64.restype = ctypes.c_uint64
_lib.o_xxhash64.argtypes = [ctypes.c_void_p, ctypes.c_size_t]
_lib.o_xxhash64_long.restype = ctypes.c_uint64
_lib.o_xxhash64_long.argtypes = [ctypes.c_int64]
_lib.o_partition_local.restype = ctypes.c_int32
_lib.o_partition_local.argtypes = [ctypes.c_int64, ctypes.c_int32]
_lib.o_partition_remote.restype = ctypes.c_int32
_lib.o_partition_remote.argtypes = [ctypes.c_int64, ctypes.c_int32]
_lib.o_hash_rows.restype = None
_lib.o_hash_rows.argtypes = [ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p,
                             ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p]
_lib.o_bigint_groupby.restype = ctypes.c_int32
_lib.o_bigint_groupby.argtypes = [ctypes.c_void_p] * 2 + [ctypes.c_int64] + [ctypes.c_void_p] * 3
_lib.o_flat_groupby.restype = ctypes.c_int32
_lib.o_flat_groupby.argtypes = [ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p,
                                ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p]
_lib.o_flat_groupby_v.restype = ctypes.c_int32
_lib.o_flat_groupby_v.argtypes = [ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p,
                                  ctypes.c_void_p]
_lib.o_grouped_sum_f64.restype = None
_lib.o_grouped_sum_f64.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_int64, ctypes.c_void_p]
_lib.o_grouped_count.restype = None
_lib.o_grouped_count.argtypes = [ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p]
_lib.o_grouped_sum_f64_exact.restype = None
_lib.o_grouped_sum_f64_exact.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                                         ctypes.c_int32, ctypes.c_void_p, ctypes.c_int64]
_lib.o_join_build_bigint.restype = ctypes.c_void_p
_lib.o_join_build_bigint.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64]
_lib.o_join_table_free.restype = None
_lib.o_join_table_free.argtypes = [ctypes.c_void_p]
_lib.o_join_table_size.restype = ctypes.c_int64
_lib.o_join_table_size.argtypes = [ctypes.c_void_p]
_lib.o_join_probe_bigint.restype = ctypes.c_int64
_lib.o_join_probe_bigint.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p]

TG_BIGINT, TG_INTEGER, TG_SMALLINT, TG_TINYINT, TG_DOUBLE, TG_DATE, TG_BOOLEAN, TG_VARCHAR = range(8)


class Q1Result(ctypes.Structure):
    _fields_ = [
        ("sum_qty", ctypes.c_double * 6), ("sum_base", ctypes.c_double * 6),
        ("sum_disc_price", ctypes.c_double * 6), ("sum_charge", ctypes.c_double * 6),
        ("sum_disc", ctypes.c_double * 6),
        ("avg_qty", ctypes.c_double * 6), ("avg_price", ctypes.c_double * 6),
        ("avg_disc", ctypes.c_double * 6),
        ("count", ctypes.c_int64 * 6),
        ("group_id_by_combo", ctypes.c_int32 * 6),
        ("n_groups", ctypes.c_int32),
    ]


_lib.o_q1_naive.restype = None
_lib.o_q1_naive.argtypes = [ctypes.c_int64] + [ctypes.c_void_p] * 7 + [ctypes.c_int32, ctypes.c_void_p]
_lib.o_q1_exact.restype = ctypes.c_double
_lib.o_q1_exact.argtypes = [ctypes.c_int64] + [ctypes.c_void_p] * 7 + [ctypes.c_int32, ctypes.c_int32, ctypes.c_void_p]


def _ptr(a):
    return None if a is None else a.ctypes.data


def lineitem_count(sf, order_start=1, order_count=None):
    if order_count is None:
        order_count = int(1_500_000 * sf)
    return _lib.tpch_lineitem_count(sf, order_start, order_count)


def gen_lineitem(sf, order_start=1, order_count=None, columns=None):
    """Generate lineitem columns. Returns dict of numpy arrays."""
    if order_count is None:
        order_count = int(1_500_000 * sf)
    cap = 7 * order_count
    all_cols = ["orderkey", "partkey", "linenumber", "shipdate", "commitdate",
                "receiptdate", "quantity", "extendedprice", "discount", "tax",
                "returnflag", "linestatus", "shipmode", "tp_cents", "suppkey",
                "shipinstruct"]
    dtypes = dict(orderkey=np.int64, partkey=np.int64, linenumber=np.int32,
                  shipdate=np.int32, commitdate=np.int32, receiptdate=np.int32,
                  quantity=np.float64, extendedprice=np.float64, discount=np.float64,
                  tax=np.float64, returnflag=np.uint8, linestatus=np.uint8,
                  shipmode=np.uint8, tp_cents=np.int64, suppkey=np.int64,
                  shipinstruct=np.uint8)
    want = set(columns) if columns else set(all_cols)
    bufs = {c: (np.empty(cap, dtypes[c]) if c in want else None) for c in all_cols}
    n = _lib.tpch_gen_lineitem(sf, order_start, order_count,
                               *[_ptr(bufs[c]) for c in all_cols])
    return {c: bufs[c][:n] for c in all_cols if bufs[c] is not None}


def gen_orders(sf, order_start=1, order_count=None):
    if order_count is None:
        order_count = int(1_500_000 * sf)
    ok = np.empty(order_count, np.int64)
    ck = np.empty(order_count, np.int64)
    od = np.empty(order_count, np.int32)
    pri = np.empty(order_count, np.uint8)
    _lib.tpch_gen_orders2(sf, order_start, order_count, _ptr(ok), _ptr(ck), _ptr(od), _ptr(pri))
    return {"orderkey": ok, "custkey": ck, "orderdate": od, "orderpriority": pri}


def gen_part(sf, part_start=1, part_count=None):
    if part_count is None:
        part_count = int(200_000 * sf)
    pk = np.empty(part_count, np.int64)
    ty = np.empty(part_count, np.uint8)
    _lib.tpch_gen_part_cols(sf, part_start, part_count, _ptr(pk), _ptr(ty))
    return {"partkey": pk, "type_id": ty.astype(np.int16)}


def gen_supplier(sf, supp_start=1, supp_count=None):
    if supp_count is None:
        supp_count = int(10_000 * sf)
    sk = np.empty(supp_count, np.int64)
    nk = np.empty(supp_count, np.uint8)
    _lib.tpch_gen_supplier(sf, supp_start, supp_count, _ptr(sk), _ptr(nk))
    return {"suppkey": sk, "nationkey": nk}


def gen_customer(sf, cust_start=1, cust_count=None):
    if cust_count is None:
        cust_count = int(150_000 * sf)
    ck = np.empty(cust_count, np.int64)
    ms = np.empty(cust_count, np.uint8)
    nk = np.empty(cust_count, np.uint8)
    ab = np.empty(cust_count, np.int64)
    _lib.tpch_gen_customer(sf, cust_start, cust_count, _ptr(ck), _ptr(ms),
                           _ptr(nk), _ptr(ab))
    return {"custkey": ck, "mktsegment": ms, "nationkey": nk,
            "acctbal_cents": ab}


def bigint_hash(v):
    return _lib.o_bigint_hash(int(v))


def double_hash(v):
    return _lib.o_double_hash(float(v))


def murmur3_mix(v):
    return _lib.o_murmur3_mix(v & 0xFFFFFFFFFFFFFFFF)


def combine_hash(prev, h):
    return _lib.o_combine_hash(prev, h)


def xxhash64(data: bytes):
    buf = np.frombuffer(data, np.uint8) if data else np.empty(0, np.uint8)
    return _lib.o_xxhash64(buf.ctypes.data if len(buf) else None, len(buf))


def xxhash64_long(v):
    return _lib.o_xxhash64_long(int(v))


def partition_local(raw_hash, count):
    return _lib.o_partition_local(raw_hash, count)


def partition_remote(raw_hash, count):
    return _lib.o_partition_remote(raw_hash, count)


_TYPE_NP = {TG_BIGINT: np.int64, TG_INTEGER: np.int32, TG_SMALLINT: np.int16,
            TG_TINYINT: np.int8, TG_DOUBLE: np.float64, TG_DATE: np.int32,
            TG_BOOLEAN: np.int8}


def hash_rows(columns, types):
    """Canonical row hash over fixed-width channels (list of np arrays)."""
    n = len(columns[0])
    out = np.empty(n, np.uint64)
    tarr = np.array(types, np.int32)
    ptrs = (ctypes.c_void_p * len(columns))(*[c.ctypes.data for c in columns])
    _lib.o_hash_rows(len(columns), tarr.ctypes.data, ctypes.addressof(ptrs), None, n, out.ctypes.data)
    return out


def bigint_groupby(keys, valid=None):
    n = len(keys)
    gids = np.empty(n, np.int32)
    vals = np.empty(max(n, 1), np.int64)
    nullg = np.zeros(1, np.int32)
    ng = _lib.o_bigint_groupby(keys.ctypes.data, _ptr(valid), n,
                               gids.ctypes.data, vals.ctypes.data, nullg.ctypes.data)
    return gids, ng, vals[:ng], int(nullg[0])


def flat_groupby(columns, types, offsets=None):
    """columns: np arrays; VARCHAR channels pass bytes array + matching entry
    in `offsets` (list parallel to columns; None for fixed-width)."""
    n = None
    for i, c in enumerate(columns):
        n = (len(offsets[i]) - 1) if (offsets and offsets[i] is not None) else len(c)
        break
    gids = np.empty(n, np.int32)
    first = np.empty(max(n, 1), np.int64)
    tarr = np.array(types, np.int32)
    ptrs = (ctypes.c_void_p * len(columns))(*[c.ctypes.data for c in columns])
    if offsets is None:
        ng = _lib.o_flat_groupby(len(columns), tarr.ctypes.data, ctypes.addressof(ptrs),
                                 n, gids.ctypes.data, first.ctypes.data)
    else:
        optrs = (ctypes.c_void_p * len(columns))(
            *[(o.ctypes.data if o is not None else None) for o in offsets])
        ng = _lib.o_flat_groupby_v(len(columns), tarr.ctypes.data, ctypes.addressof(ptrs),
                                   ctypes.addressof(optrs), n, gids.ctypes.data,
                                   first.ctypes.data)
    return gids, ng, first[:ng]


def grouped_sum_f64(gids, vals, n_groups, valid=None):
    out = np.zeros(n_groups, np.float64)
    _lib.o_grouped_sum_f64(gids.ctypes.data, vals.ctypes.data, _ptr(valid), len(gids), out.ctypes.data)
    return out


def grouped_count(gids, n_groups):
    out = np.zeros(n_groups, np.int64)
    _lib.o_grouped_count(gids.ctypes.data, len(gids), out.ctypes.data)
    return out


def grouped_sum_f64_exact(gids, vals, n_groups, scale_pow=43):
    out = np.zeros(n_groups, np.float64)
    _lib.o_grouped_sum_f64_exact(gids.ctypes.data, vals.ctypes.data, len(gids),
                                 scale_pow, out.ctypes.data, n_groups)
    return out


class JoinTable:
    def __init__(self, keys, valid=None, positions_per_page=None):
        n = len(keys)
        if positions_per_page is None:
            positions_per_page = max(n, 1)
        self._h = _lib.o_join_build_bigint(keys.ctypes.data, _ptr(valid), n, positions_per_page)
        self.n = n

    def probe(self, probe_keys, probe_valid=None, cap=None):
        m = len(probe_keys)
        if cap is None:
            cap = max(4 * m + 16, 1024)
        op = np.empty(cap, np.int32)
        ob = np.empty(cap, np.int32)
        cnt = _lib.o_join_probe_bigint(self._h, probe_keys.ctypes.data, _ptr(probe_valid),
                                       m, cap, op.ctypes.data, ob.ctypes.data)
        assert cnt >= 0, "probe capacity exceeded"
        return op[:cnt], ob[:cnt]

    def table_size(self):
        return _lib.o_join_table_size(self._h)

    def __del__(self):
        try:
            _lib.o_join_table_free(self._h)
        except Exception:
            pass


def q1_naive(cols, cutoff=10471):
    r = Q1Result()
    _lib.o_q1_naive(len(cols["shipdate"]), cols["shipdate"].ctypes.data,
                    cols["quantity"].ctypes.data, cols["extendedprice"].ctypes.data,
                    cols["discount"].ctypes.data, cols["tax"].ctypes.data,
                    cols["returnflag"].ctypes.data, cols["linestatus"].ctypes.data,
                    cutoff, ctypes.byref(r))
    return r


def q1_exact(cols, cutoff=10471, threads=1):
    r = Q1Result()
    elapsed = _lib.o_q1_exact(len(cols["shipdate"]), cols["shipdate"].ctypes.data,
                              cols["quantity"].ctypes.data, cols["extendedprice"].ctypes.data,
                              cols["discount"].ctypes.data, cols["tax"].ctypes.data,
                              cols["returnflag"].ctypes.data, cols["linestatus"].ctypes.data,
                              cutoff, threads, ctypes.byref(r))
    return r, elapsed


# ---- round 2: extended tables + text (tpch_text.h restatement) ----
_lib.tpch_gen_part2.restype = None
_lib.tpch_gen_part2.argtypes = [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 7
_lib.tpch_gen_partsupp2.restype = None
_lib.tpch_gen_partsupp2.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 4
_lib.tpch_gen_supplier2.restype = None
_lib.tpch_gen_supplier2.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 3
_lib.tpch_gen_orders3.restype = None
_lib.tpch_gen_orders3.argtypes = [ctypes.c_double, ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 9
_lib.tpch_text_pool.restype = ctypes.c_void_p
_lib.tpch_gen_supplier_comments.restype = None
_lib.tpch_gen_supplier_comments.argtypes = [ctypes.c_int64, ctypes.c_int64,
                                            ctypes.c_void_p, ctypes.c_int32,
                                            ctypes.c_void_p]


def gen_part2(sf, part_start=1, part_count=None):
    if part_count is None:
        part_count = int(200_000 * sf)
    pk = np.empty(part_count, np.int64)
    ty = np.empty(part_count, np.int16)
    br = np.empty(part_count, np.uint8)
    sz = np.empty(part_count, np.int32)
    cn = np.empty(part_count, np.uint8)
    nm = np.empty(part_count * 5, np.uint8)
    rp = np.empty(part_count, np.int64)
    _lib.tpch_gen_part2(part_start, part_count, _ptr(pk), _ptr(ty), _ptr(br),
                        _ptr(sz), _ptr(cn), _ptr(nm), _ptr(rp))
    return {"partkey": pk, "type_id": ty, "brand": br, "size": sz,
            "container": cn, "name_ids": nm.reshape(part_count, 5),
            "retail_cents": rp}


def gen_partsupp(sf, part_start=1, part_count=None):
    if part_count is None:
        part_count = int(200_000 * sf)
    n = part_count * 4
    pk = np.empty(n, np.int64)
    sk = np.empty(n, np.int64)
    aq = np.empty(n, np.int32)
    sc = np.empty(n, np.int64)
    _lib.tpch_gen_partsupp2(sf, part_start, part_count, _ptr(pk), _ptr(sk),
                            _ptr(aq), _ptr(sc))
    return {"partkey": pk, "suppkey": sk, "availqty": aq,
            "supplycost_cents": sc}


def gen_supplier2(sf, supp_start=1, supp_count=None):
    if supp_count is None:
        supp_count = int(10_000 * sf)
    sk = np.empty(supp_count, np.int64)
    nk = np.empty(supp_count, np.uint8)
    ab = np.empty(supp_count, np.int64)
    _lib.tpch_gen_supplier2(sf, supp_start, supp_count, _ptr(sk), _ptr(nk), _ptr(ab))
    return {"suppkey": sk, "nationkey": nk, "acctbal_cents": ab}


def gen_orders3(sf, order_start=1, order_count=None):
    if order_count is None:
        order_count = int(1_500_000 * sf)
    ok = np.empty(order_count, np.int64)
    ck = np.empty(order_count, np.int64)
    od = np.empty(order_count, np.int32)
    pri = np.empty(order_count, np.uint8)
    st = np.empty(order_count, np.uint8)
    tp = np.empty(order_count, np.int64)
    co = np.empty(order_count, np.int64)
    cl = np.empty(order_count, np.int32)
    cln = np.empty(order_count, np.int32)
    _lib.tpch_gen_orders3(sf, order_start, order_count, _ptr(ok), _ptr(ck),
                          _ptr(od), _ptr(pri), _ptr(st), _ptr(tp), _ptr(co),
                          _ptr(cln), _ptr(cl))
    return {"orderkey": ok, "custkey": ck, "orderdate": od, "orderpriority": pri,
            "orderstatus": st, "totalprice_cents": tp, "cmnt_off": co,
            "cmnt_len": cln, "clerk": cl}


def text_pool_bytes():
    """The full 300 MiB pool as a (read-only) numpy uint8 view."""
    p = _lib.tpch_text_pool()
    return np.ctypeslib.as_array((ctypes.c_uint8 * (300 * 1024 * 1024)).from_address(p))


def gen_supplier_comments(supp_start, count):
    buf = np.zeros((count, 101), np.uint8)
    lens = np.zeros(count, np.int32)
    _lib.tpch_gen_supplier_comments(supp_start, count, _ptr(buf), 101, _ptr(lens))
    return [bytes(buf[i, :lens[i]]).decode("latin1") for i in range(count)]


NATIONS = ["ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
           "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ",
           "JAPAN", "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU",
           "CHINA", "ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA",
           "UNITED KINGDOM", "UNITED STATES"]
NATION_REGION = [0, 1, 1, 1, 4, 0, 3, 3, 2, 2, 4, 4, 2, 4, 0, 0, 0, 1, 2, 3,
                 4, 2, 3, 3, 1]
REGIONS = ["AFRICA", "AMERICA", "ASIA", "EUROPE", "MIDDLE EAST"]
COLOR_GREEN = 33
COLOR_FOREST = 28
