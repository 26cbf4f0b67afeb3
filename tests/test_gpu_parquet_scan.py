"""Parquet -> HBM -> fused Q1: the "SF-Parquet" scan shape (config 2) at test
scale. The file is written on the box by pyarrow from oracle-generated rows,
decoded by the native reader, uploaded as flat columns, and the fused Q1
result is compared bit-exactly against the oracle exact leg.
"""
import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

import oracle  # noqa: E402

pytestmark = pytest.mark.gpu


def test_parquet_scan_to_q1(tmp_path):
    import trino_amd
    from trino_amd.parquet import ParquetFile

    li = oracle.gen_lineitem(0.01)
    flags = np.array(["A", "N", "R"])
    stat = np.array(["F", "O"])
    f = tmp_path / "lineitem.parquet"
    pq.write_table(pa.table({
        "l_shipdate": li["shipdate"], "l_quantity": li["quantity"],
        "l_extendedprice": li["extendedprice"], "l_discount": li["discount"],
        "l_tax": li["tax"], "l_returnflag": flags[li["returnflag"]],
        "l_linestatus": stat[li["linestatus"]],
    }), f, compression="snappy", data_page_version="1.0", row_group_size=16384)

    pf = ParquetFile(f)
    n = pf.num_rows
    sd, _, _ = pf.read_column("l_shipdate")
    qt, _, _ = pf.read_column("l_quantity")
    ep, _, _ = pf.read_column("l_extendedprice")
    di, _, _ = pf.read_column("l_discount")
    tx, _, _ = pf.read_column("l_tax")
    rf_ids, _, rf_dict = pf.read_column("l_returnflag")
    ls_ids, _, ls_dict = pf.read_column("l_linestatus")
    pf.close()
    # remap parquet dictionary ids to the engine's sorted dictionary ids
    rf_map = np.array([b"ANR".index(d) for d in rf_dict], np.uint8)
    ls_map = np.array([b"FO".index(d) for d in ls_dict], np.uint8)
    rf = rf_map[rf_ids]
    ls = ls_map[ls_ids]
    assert np.array_equal(rf, li["returnflag"]) and np.array_equal(ls, li["linestatus"])

    # upload decoded columns and run the fused Q1 through the C ABI
    s = trino_amd.Session(0)
    try:
        import ctypes
        from trino_amd import tpch_queries, LineitemCols, _lib, _check
        bufs = {}
        for name, arr in (("shipdate", sd), ("quantity", qt), ("extendedprice", ep),
                          ("discount", di), ("tax", tx), ("returnflag", rf),
                          ("linestatus", ls)):
            p = tpch_queries._device_buffer(s, arr.nbytes)
            _check(_lib.tg_copy_htod(s._h, p, arr.ctypes.data, arr.nbytes))
            bufs[name] = p
        cols = LineitemCols()
        cols.row_count = n
        for name in bufs:
            setattr(cols, name, bufs[name].value)
        gpu = s.q1(cols)
        ref, _ = oracle.q1_exact(li)
        for c in range(6):
            assert gpu.count[c] == ref.count[c]
            assert gpu.sum_base[c] == ref.sum_base[c]
            assert gpu.sum_disc_price[c] == ref.sum_disc_price[c]
        for p in bufs.values():
            tpch_queries._device_free(s, p)
    finally:
        s.close()
