"""trino_amd.parquet — ctypes plumbing over the native Parquet reader
(trino_amd/csrc/parquet.cpp; §8f row 1). Host-side decode into flat column
buffers (the same layouts the Block mirror uses), ready for flat HtoD upload.
Pure host C++ under the hood — usable (and tested) without a GPU.
"""
import ctypes

import numpy as np

from . import _lib, _check

PQ_INT32, PQ_INT64, PQ_DOUBLE, PQ_BYTE_ARRAY = 1, 2, 5, 6

_lib.tg_parquet_open.restype = ctypes.c_int
_lib.tg_parquet_open.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_void_p]
_lib.tg_parquet_close.restype = None
_lib.tg_parquet_close.argtypes = [ctypes.c_void_p]
_lib.tg_parquet_num_rows.restype = ctypes.c_int64
_lib.tg_parquet_num_rows.argtypes = [ctypes.c_void_p]
_lib.tg_parquet_num_columns.restype = ctypes.c_int32
_lib.tg_parquet_num_columns.argtypes = [ctypes.c_void_p]
_lib.tg_parquet_column_name.restype = ctypes.c_char_p
_lib.tg_parquet_column_name.argtypes = [ctypes.c_void_p, ctypes.c_int32]
_lib.tg_parquet_physical_type.restype = ctypes.c_int32
_lib.tg_parquet_physical_type.argtypes = [ctypes.c_void_p, ctypes.c_int32]
_lib.tg_parquet_read_column.restype = ctypes.c_int
_lib.tg_parquet_read_column.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                                        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p,
                                        ctypes.c_void_p]

_PT_NP = {PQ_INT32: np.int32, PQ_INT64: np.int64, PQ_DOUBLE: np.float64}


class ParquetFile:
    def __init__(self, path, session=None):
        self._h = ctypes.c_void_p()
        sh = session._h if session is not None else None
        _check(_lib.tg_parquet_open(sh, str(path).encode(), ctypes.byref(self._h)))
        self.num_rows = _lib.tg_parquet_num_rows(self._h)
        self.columns = [_lib.tg_parquet_column_name(self._h, i).decode()
                        for i in range(_lib.tg_parquet_num_columns(self._h))]
        self.types = [_lib.tg_parquet_physical_type(self._h, i)
                      for i in range(len(self.columns))]

    def read_column(self, col, session=None):
        """Returns (values or ids, valid_bitmap_or_None, dictionary_or_None).
        Numeric columns: (np array, valid, None). BYTE_ARRAY: (int32 ids,
        valid, list[bytes] dictionary)."""
        if isinstance(col, str):
            col = self.columns.index(col)
        pt = self.types[col]
        n = self.num_rows
        valid = np.full((n + 63) // 64, ~np.uint64(0), np.uint64)
        sh = session._h if session is not None else None
        if pt == PQ_BYTE_ARRAY:
            ids = np.empty(n, np.int32)
            cap = 1 << 24
            dbytes = np.empty(cap, np.uint8)
            doffs = np.empty(1 << 20, np.int32)
            dcount = ctypes.c_int32(0)
            _check(_lib.tg_parquet_read_column(sh, self._h, col, None,
                                               valid.ctypes.data, ids.ctypes.data,
                                               dbytes.ctypes.data, cap,
                                               doffs.ctypes.data, ctypes.byref(dcount)))
            dc = dcount.value
            dictionary = [bytes(dbytes[doffs[i]:doffs[i + 1]]) for i in range(dc)]
            return ids, _maybe_valid(valid, n), dictionary
        vals = np.empty(n, _PT_NP[pt])
        _check(_lib.tg_parquet_read_column(sh, self._h, col, vals.ctypes.data,
                                           valid.ctypes.data, None, None, 0, None, None))
        return vals, _maybe_valid(valid, n), None

    def close(self):
        if self._h:
            _lib.tg_parquet_close(self._h)
            self._h = None


def _maybe_valid(valid, n):
    full = np.full((n + 63) // 64, ~np.uint64(0), np.uint64)
    if n % 64:
        full[-1] = np.uint64((1 << (n % 64)) - 1)
        got = valid.copy()
        got[-1] &= full[-1]
    else:
        got = valid
    return None if np.array_equal(got, full) else valid


_lib.tg_parquet_read_columns.restype = ctypes.c_int
_lib.tg_parquet_read_columns.argtypes = [ctypes.c_void_p] * 2 + \
    [ctypes.c_void_p, ctypes.c_int32] + [ctypes.c_void_p] * 7


def read_columns(pf, names, session=None):
    """Parallel multi-column decode (columns x row groups): returns
    {name: ndarray or (ids, dict_strings)} like ParquetFile.read_column."""
    import numpy as np
    sh = session._h if session is not None else None
    n = pf.num_rows
    k = len(names)
    cols = [pf.columns.index(x) for x in names]
    types = [pf.types[c] for c in cols]
    vals_ptr = (ctypes.c_void_p * k)()
    ids_ptr = (ctypes.c_void_p * k)()
    db_ptr = (ctypes.c_void_p * k)()
    do_ptr = (ctypes.c_void_p * k)()
    dc = np.zeros(k, np.int32)
    dc_ptr = (ctypes.c_void_p * k)()
    caps = (ctypes.c_int64 * k)()
    keep = {}
    for i, (c, t) in enumerate(zip(cols, types)):
        if t == 6:       # BYTE_ARRAY: dictionary ids + global dict
            ids = np.empty(n, np.int32)
            db = np.empty(1 << 24, np.uint8)
            do = np.empty(1 << 20, np.int32)
            keep[i] = (ids, db, do)
            ids_ptr[i] = ids.ctypes.data
            db_ptr[i] = db.ctypes.data
            do_ptr[i] = do.ctypes.data
            caps[i] = db.nbytes
            dc_ptr[i] = dc.ctypes.data + 4 * i
        else:
            dt = {1: np.int32, 2: np.int64, 5: np.float64}[t]
            a = np.empty(n, dt)
            keep[i] = a
            vals_ptr[i] = a.ctypes.data
    carr = (ctypes.c_int32 * k)(*cols)
    _check(_lib.tg_parquet_read_columns(sh, pf._h, carr, k, vals_ptr, None,
                                        ids_ptr, db_ptr, caps, do_ptr, dc_ptr))
    out = {}
    for i, (name, t) in enumerate(zip(names, types)):
        if t == 6:
            ids, db, do = keep[i]
            cnt = int(dc[i])
            strs = [bytes(db[do[j]:do[j + 1]]) for j in range(cnt)]
            out[name] = (ids, strs)
        else:
            out[name] = keep[i]
    return out
