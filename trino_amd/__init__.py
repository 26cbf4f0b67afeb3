"""trino_amd — MI355X-native implementation of Trino's columnar operator hot
path. Product path: ctypes over the in-tree C-ABI library
``libtrino_gpu.so`` (include/trino_gpu.h). There is NO CPU fallback: loading
works anywhere (for symbol checks), but creating a session requires a real
HIP device and fails loudly otherwise.
"""
import ctypes
import os

__version__ = "0.1"

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libtrino_gpu.so")


class TrinoGpuError(RuntimeError):
    pass


def _load():
    if not os.path.exists(_SO):
        raise TrinoGpuError(
            f"{_SO} missing — build it with __graft_entry__.build() / "
            "`make -C trino_amd/csrc` (the product path never falls back to CPU)")
    return ctypes.CDLL(_SO)


_lib = _load()

_lib.tg_last_error.restype = ctypes.c_char_p
_lib.tg_version.restype = ctypes.c_char_p
_lib.tg_session_create.restype = ctypes.c_int
_lib.tg_session_create.argtypes = [ctypes.c_int, ctypes.c_void_p]
_lib.tg_session_close.restype = None
_lib.tg_session_close.argtypes = [ctypes.c_void_p]


class LineitemCols(ctypes.Structure):
    _fields_ = [
        ("row_count", ctypes.c_int64),
        ("orderkey", ctypes.c_void_p),
        ("shipdate", ctypes.c_void_p),
        ("quantity", ctypes.c_void_p),
        ("extendedprice", ctypes.c_void_p),
        ("discount", ctypes.c_void_p),
        ("tax", ctypes.c_void_p),
        ("returnflag", ctypes.c_void_p),
        ("linestatus", ctypes.c_void_p),
        ("commitdate", ctypes.c_void_p),
        ("receiptdate", ctypes.c_void_p),
        ("partkey", ctypes.c_void_p),
        ("shipmode", ctypes.c_void_p),
        ("tp_cents", ctypes.c_void_p),
        ("suppkey", ctypes.c_void_p),
        ("shipinstruct", ctypes.c_void_p),
    ]


class Q1Result(ctypes.Structure):
    _fields_ = [
        ("sum_qty", ctypes.c_double * 6), ("sum_base", ctypes.c_double * 6),
        ("sum_disc_price", ctypes.c_double * 6), ("sum_charge", ctypes.c_double * 6),
        ("avg_qty", ctypes.c_double * 6), ("avg_price", ctypes.c_double * 6),
        ("avg_disc", ctypes.c_double * 6), ("sum_disc", ctypes.c_double * 6),
        ("count", ctypes.c_int64 * 6),
        ("elapsed_ms", ctypes.c_double),
        ("raw", ctypes.c_uint64 * 60),
    ]


_lib.tg_tpch_lineitem_alloc.restype = ctypes.c_int
_lib.tg_tpch_lineitem_alloc.argtypes = [ctypes.c_void_p, ctypes.c_double,
                                        ctypes.c_int64, ctypes.c_int64,
                                        ctypes.c_int, ctypes.POINTER(LineitemCols)]
_lib.tg_tpch_lineitem_free.restype = ctypes.c_int
_lib.tg_tpch_lineitem_free.argtypes = [ctypes.c_void_p, ctypes.POINTER(LineitemCols)]
_lib.tg_q1_run.restype = ctypes.c_int
_lib.tg_q1_run.argtypes = [ctypes.c_void_p, ctypes.POINTER(LineitemCols),
                           ctypes.c_int32, ctypes.POINTER(Q1Result)]
_lib.tg_q1_run_naive.restype = ctypes.c_int
_lib.tg_q1_run_naive.argtypes = [ctypes.c_void_p, ctypes.POINTER(LineitemCols),
                                 ctypes.c_int32, ctypes.POINTER(Q1Result)]
_lib.tg_tpch_gen_orders.restype = ctypes.c_int
_lib.tg_tpch_gen_orders.argtypes = [ctypes.c_void_p, ctypes.c_double, ctypes.c_int64,
                                    ctypes.c_int64] + [ctypes.c_void_p] * 4
_lib.tg_tpch_gen_customer.restype = ctypes.c_int
_lib.tg_tpch_gen_customer.argtypes = [ctypes.c_void_p, ctypes.c_double, ctypes.c_int64,
                                      ctypes.c_int64] + [ctypes.c_void_p] * 4
_lib.tg_tpch_gen_supplier.restype = ctypes.c_int
_lib.tg_tpch_gen_supplier.argtypes = [ctypes.c_void_p, ctypes.c_double, ctypes.c_int64,
                                      ctypes.c_int64] + [ctypes.c_void_p] * 2


def _check(status):
    if status != 0:
        raise TrinoGpuError(f"status {status}: {_lib.tg_last_error().decode()}")


def version():
    return _lib.tg_version().decode()


class Session:
    """Device context (one per process per GPU)."""

    def __init__(self, device=0):
        self._h = ctypes.c_void_p()
        _check(_lib.tg_session_create(device, ctypes.byref(self._h)))
        self.device = device

    def close(self):
        if self._h:
            _lib.tg_session_close(self._h)
            self._h = None

    # --- TPC-H device generator (bench/test input infrastructure) ---
    def tpch_lineitem(self, sf, order_start=1, order_count=None, with_orderkey=False,
                      with_dates=False, with_partkey=False, with_shipmode=False,
                      with_totalprice=False, with_suppkey=False,
                      with_shipinstruct=False):
        if order_count is None:
            order_count = int(1_500_000 * sf)
        cols = LineitemCols()
        flags = ((1 if with_orderkey else 0) | (2 if with_dates else 0) |
                 (4 if with_partkey else 0) | (8 if with_shipmode else 0) |
                 (16 if with_totalprice else 0) | (32 if with_suppkey else 0) |
                 (64 if with_shipinstruct else 0))
        _check(_lib.tg_tpch_lineitem_alloc(self._h, sf, order_start, order_count,
                                           flags, ctypes.byref(cols)))
        return cols

    def tpch_lineitem_free(self, cols):
        _check(_lib.tg_tpch_lineitem_free(self._h, ctypes.byref(cols)))

    # --- fused Q1 (the north-star benchmark path) ---
    def q1(self, cols, cutoff=10471):
        r = Q1Result()
        _check(_lib.tg_q1_run(self._h, ctypes.byref(cols), cutoff, ctypes.byref(r)))
        return r

    def q1_naive(self, cols, cutoff=10471):
        r = Q1Result()
        _check(_lib.tg_q1_run_naive(self._h, ctypes.byref(cols), cutoff, ctypes.byref(r)))
        return r


_lib.tg_copy_dtoh.restype = ctypes.c_int
_lib.tg_copy_dtoh.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64]


def copy_dtoh(session, host_array, dev_ptr, bytes_=None):
    """Test helper: copy a device buffer into a numpy array."""
    if bytes_ is None:
        bytes_ = host_array.nbytes
    _check(_lib.tg_copy_dtoh(session._h, host_array.ctypes.data, dev_ptr, bytes_))
    return host_array


def lineitem_to_host(session, cols):
    """Test helper: pull generated device columns to numpy (parity checks)."""
    import numpy as np
    n = cols.row_count
    out = {}
    spec = [("shipdate", np.int32), ("quantity", np.float64),
            ("extendedprice", np.float64), ("discount", np.float64),
            ("tax", np.float64), ("returnflag", np.uint8), ("linestatus", np.uint8)]
    if cols.orderkey:
        spec.append(("orderkey", np.int64))
    if cols.shipmode:
        spec.append(("shipmode", np.uint8))
    for name, dt in spec:
        a = np.empty(n, dt)
        copy_dtoh(session, a, getattr(cols, name))
        out[name] = a
    return out


_lib.tg_copy_htod.restype = ctypes.c_int
_lib.tg_copy_htod.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64]


_lib.tg_tpch_gen_part.restype = ctypes.c_int
_lib.tg_tpch_gen_part.argtypes = [ctypes.c_void_p, ctypes.c_double, ctypes.c_int64,
                                  ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p]


_lib.tg_session_memory.restype = ctypes.c_int
_lib.tg_session_memory.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p]


def session_memory(session):
    """(total_pooled_bytes, cached_bytes) — live device use = total - cached."""
    tot = ctypes.c_int64()
    cach = ctypes.c_int64()
    _check(_lib.tg_session_memory(session._h, ctypes.byref(tot), ctypes.byref(cach)))
    return tot.value, cach.value
