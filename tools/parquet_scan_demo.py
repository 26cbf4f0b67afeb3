#!/usr/bin/env python3
"""Parquet scan demo (BASELINE config 2 shape at demo scale): write TPC-H
lineitem SF<n> to Parquet (pyarrow, snappy), decode it with the NATIVE reader
(trino_amd/csrc/parquet.cpp), upload the flat columns to HBM, run the fused
Q1 kernel, and verify bit-exactly against the oracle. Reports decode and
end-to-end rates. Run on a GPU box: python tools/parquet_scan_demo.py [sf]
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pyarrow as pa
import pyarrow.parquet as pq

import oracle
import trino_amd
from trino_amd import tpch_queries
from trino_amd.parquet import ParquetFile


def main():
    sf = float(sys.argv[1]) if len(sys.argv) > 1 else 10.0
    path = os.environ.get("TMPDIR", "/tmp") + f"/lineitem_sf{sf:g}.parquet"
    t0 = time.time()
    li = oracle.gen_lineitem(sf)
    n = len(li["shipdate"])
    print(f"generated {n:,} rows (oracle, CPU) in {time.time()-t0:.1f}s", flush=True)
    flags = np.array(["A", "N", "R"])
    stat = np.array(["F", "O"])
    t0 = time.time()
    pq.write_table(pa.table({
        "l_shipdate": li["shipdate"], "l_quantity": li["quantity"],
        "l_extendedprice": li["extendedprice"], "l_discount": li["discount"],
        "l_tax": li["tax"], "l_returnflag": flags[li["returnflag"]],
        "l_linestatus": stat[li["linestatus"]],
    }), path, compression="snappy", data_page_version="1.0",
        row_group_size=1 << 20)
    sz = os.path.getsize(path)
    print(f"pyarrow wrote {sz/2**20:.0f} MiB in {time.time()-t0:.1f}s", flush=True)

    # ---- native decode (host) ----
    t0 = time.time()
    pf = ParquetFile(path)
    cols = {}
    for name in ("l_shipdate", "l_quantity", "l_extendedprice", "l_discount", "l_tax"):
        cols[name], _, _ = pf.read_column(name)
    rf_ids, _, rf_dict = pf.read_column("l_returnflag")
    ls_ids, _, ls_dict = pf.read_column("l_linestatus")
    pf.close()
    t_dec = time.time() - t0
    raw_bytes = n * 38
    print(f"native decode: {t_dec:.2f}s = {sz/2**20/t_dec:.0f} MiB/s compressed, "
          f"{raw_bytes/2**20/t_dec:.0f} MiB/s of flat columns", flush=True)
    rf = np.array([b"ANR".index(d) for d in rf_dict], np.uint8)[rf_ids]
    ls = np.array([b"FO".index(d) for d in ls_dict], np.uint8)[ls_ids]

    # ---- upload + fused Q1 ----
    s = trino_amd.Session(0)
    try:
        t0 = time.time()
        bufs = {}
        import ctypes
        from trino_amd import LineitemCols, _lib, _check
        for name, arr in (("shipdate", cols["l_shipdate"]), ("quantity", cols["l_quantity"]),
                          ("extendedprice", cols["l_extendedprice"]),
                          ("discount", cols["l_discount"]), ("tax", cols["l_tax"]),
                          ("returnflag", rf), ("linestatus", ls)):
            p = tpch_queries._device_buffer(s, arr.nbytes)
            _check(_lib.tg_copy_htod(s._h, p, arr.ctypes.data, arr.nbytes))
            bufs[name] = p
        t_up = time.time() - t0
        lc = LineitemCols()
        lc.row_count = n
        for name in bufs:
            setattr(lc, name, bufs[name].value)
        r = s.q1(lc)
        ref, _ = oracle.q1_exact(li)
        okc = all(r.count[c] == ref.count[c] and r.sum_base[c] == ref.sum_base[c]
                  and r.sum_charge[c] == ref.sum_charge[c] for c in range(6))
        print(f"upload {raw_bytes/2**30:.1f} GiB in {t_up:.2f}s; fused Q1 kernel "
              f"{r.elapsed_ms:.3f} ms; parity vs oracle exact: "
              f"{'BIT-EXACT' if okc else 'MISMATCH'}", flush=True)
        for p in bufs.values():
            tpch_queries._device_free(s, p)
        if not okc:
            sys.exit(1)
    finally:
        s.close()
    os.unlink(path)


if __name__ == "__main__":
    main()
