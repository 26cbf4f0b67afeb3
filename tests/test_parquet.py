"""Native Parquet reader tests (CPU — host decode; no GPU needed).

Fixtures are written at test time with pyarrow (available in both the build
and GPU images) and decoded by trino_amd/csrc/parquet.cpp; parity = the
original arrays. Covers PLAIN + RLE_DICTIONARY, v1 data pages, snappy/zstd/
uncompressed, optional columns with nulls, BYTE_ARRAY dictionaries, and the
TPC-H lineitem Q1 column set produced by the oracle generator.
"""
import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

import oracle  # noqa: E402


def _reader(path):
    from trino_amd.parquet import ParquetFile
    return ParquetFile(path)


@pytest.mark.parametrize("compression", ["none", "snappy", "zstd"])
def test_numeric_roundtrip(tmp_path, compression):
    r = np.random.default_rng(0)
    n = 50_000
    t = pa.table({
        "a": r.integers(-2**60, 2**60, n),
        "b": r.integers(-2**30, 2**30, n).astype(np.int32),
        "c": r.standard_normal(n),
    })
    f = tmp_path / f"t_{compression}.parquet"
    pq.write_table(t, f, compression=compression, data_page_version="1.0")
    pf = _reader(f)
    assert pf.num_rows == n
    a, av, _ = pf.read_column("a")
    b, bv, _ = pf.read_column("b")
    c, cv, _ = pf.read_column("c")
    assert np.array_equal(a, t["a"].to_numpy())
    assert np.array_equal(b, t["b"].to_numpy())
    assert np.array_equal(c, t["c"].to_numpy())
    assert av is None and bv is None and cv is None
    pf.close()


def test_dictionary_encoded_numeric(tmp_path):
    """pyarrow dictionary-encodes low-cardinality numerics by default."""
    r = np.random.default_rng(1)
    n = 100_000
    vals = r.integers(0, 50, n)                  # low cardinality -> dict pages
    f = tmp_path / "dict.parquet"
    pq.write_table(pa.table({"v": vals}), f, compression="snappy",
                   data_page_version="1.0")
    pf = _reader(f)
    v, valid, _ = pf.read_column("v")
    assert np.array_equal(v, vals)
    pf.close()


def test_nulls(tmp_path):
    n = 10_000
    vals = np.arange(n, dtype=np.int64).astype(object)
    for i in range(0, n, 7):
        vals[i] = None
    f = tmp_path / "nulls.parquet"
    pq.write_table(pa.table({"v": pa.array(vals, pa.int64())}), f,
                   compression="none", data_page_version="1.0")
    pf = _reader(f)
    v, valid, _ = pf.read_column("v")
    assert valid is not None
    for i in range(n):
        isvalid = bool((valid[i // 64] >> np.uint64(i % 64)) & np.uint64(1))
        assert isvalid == (vals[i] is not None)
        if isvalid:
            assert v[i] == vals[i]
        else:
            assert v[i] == 0
    pf.close()


def test_byte_array_dictionary(tmp_path):
    n = 20_000
    r = np.random.default_rng(2)
    flags = np.array(["A", "N", "R"])[r.integers(0, 3, n)]
    f = tmp_path / "ba.parquet"
    pq.write_table(pa.table({"flag": flags}), f, compression="zstd",
                   data_page_version="1.0")
    pf = _reader(f)
    ids, valid, dictionary = pf.read_column("flag")
    decoded = np.array([dictionary[i].decode() for i in ids])
    assert np.array_equal(decoded, flags)
    pf.close()


def test_multi_row_group(tmp_path):
    n = 200_000
    vals = np.arange(n, dtype=np.int64) * 3
    f = tmp_path / "rg.parquet"
    pq.write_table(pa.table({"v": vals}), f, compression="snappy",
                   row_group_size=30_000, data_page_version="1.0")
    pf = _reader(f)
    v, _, _ = pf.read_column("v")
    assert np.array_equal(v, vals)
    pf.close()


def test_lineitem_q1_columns(tmp_path):
    """the SF-Parquet scan shape: oracle-generated lineitem written to
    parquet, decoded natively, compared bit-exactly"""
    li = oracle.gen_lineitem(0.01)
    names = ["l_shipdate", "l_quantity", "l_extendedprice", "l_discount",
             "l_tax", "l_returnflag", "l_linestatus"]
    flags = np.array(["A", "N", "R"])
    stat = np.array(["F", "O"])
    t = pa.table({
        "l_shipdate": li["shipdate"],
        "l_quantity": li["quantity"],
        "l_extendedprice": li["extendedprice"],
        "l_discount": li["discount"],
        "l_tax": li["tax"],
        "l_returnflag": flags[li["returnflag"]],
        "l_linestatus": stat[li["linestatus"]],
    })
    f = tmp_path / "lineitem.parquet"
    pq.write_table(t, f, compression="snappy", data_page_version="1.0")
    pf = _reader(f)
    assert pf.num_rows == len(li["shipdate"])
    sd, _, _ = pf.read_column("l_shipdate")
    qt, _, _ = pf.read_column("l_quantity")
    ep, _, _ = pf.read_column("l_extendedprice")
    assert np.array_equal(sd, li["shipdate"])
    assert np.array_equal(qt, li["quantity"])
    assert np.array_equal(ep, li["extendedprice"])
    ids, _, dictionary = pf.read_column("l_returnflag")
    decoded = np.array([dictionary[i] for i in ids])
    assert np.array_equal(decoded, flags[li["returnflag"]].astype("S1"))
    pf.close()
