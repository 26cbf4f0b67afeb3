"""GPU parity tests for the fused Q1 path and the device TPC-H generator.

All tests call through the C-ABI (trino_amd ctypes over libtrino_gpu.so);
the oracle is the checker only. Tolerances:
 - device generator vs CPU oracle: bit-exact;
 - fused kernel vs oracle exact leg: bit-exact (error-free fixed-point sums);
 - parity-mode sequential kernel vs oracle naive leg: bit-exact.
"""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def sess():
    import trino_amd
    s = trino_amd.Session(0)
    yield s
    s.close()


def assert_q1_equal(gpu, ora, check_avg=True):
    for c in range(6):
        assert gpu.count[c] == ora.count[c], ("count", c, gpu.count[c], ora.count[c])
        assert gpu.sum_qty[c] == ora.sum_qty[c], ("sum_qty", c)
        assert gpu.sum_base[c] == ora.sum_base[c], ("sum_base", c, gpu.sum_base[c], ora.sum_base[c])
        assert gpu.sum_disc_price[c] == ora.sum_disc_price[c], ("sum_disc_price", c)
        assert gpu.sum_charge[c] == ora.sum_charge[c], ("sum_charge", c)
        assert gpu.sum_disc[c] == ora.sum_disc[c], ("sum_disc", c)
        if check_avg and ora.count[c]:
            assert gpu.avg_qty[c] == ora.avg_qty[c]
            assert gpu.avg_price[c] == ora.avg_price[c]
            assert gpu.avg_disc[c] == ora.avg_disc[c]


class TestDeviceGenerator:
    @pytest.mark.parametrize("sf,start,count", [
        (0.01, 1, 15000),      # full sf0.01
        (1.0, 700_001, 5000),  # mid-table part (pins device seek)
        (1.0, 1_499_001, 1000),  # table tail
    ])
    def test_bitwise_vs_oracle(self, sess, sf, start, count):
        import trino_amd
        cols = sess.tpch_lineitem(sf, start, count, with_orderkey=True)
        dev = trino_amd.lineitem_to_host(sess, cols)
        ref = oracle.gen_lineitem(sf, start, count)
        assert cols.row_count == len(ref["shipdate"])
        for k in ("orderkey", "shipdate", "quantity", "extendedprice",
                  "discount", "tax", "returnflag", "linestatus"):
            assert np.array_equal(dev[k], ref[k]), k
        sess.tpch_lineitem_free(cols)

    def test_rowcount_sf001(self, sess):
        cols = sess.tpch_lineitem(0.01)
        assert cols.row_count == 60175   # sf0.01 statistics fixture
        sess.tpch_lineitem_free(cols)


class TestQ1Fused:
    def test_sf001_exact(self, sess):
        cols = sess.tpch_lineitem(0.01)
        gpu = sess.q1(cols)
        ref, _ = oracle.q1_exact(oracle.gen_lineitem(0.01))
        assert_q1_equal(gpu, ref)
        sess.tpch_lineitem_free(cols)

    def test_sf1_exact_and_answer(self, sess):
        cols = sess.tpch_lineitem(1.0)
        gpu = sess.q1(cols)
        ref, _ = oracle.q1_exact(oracle.gen_lineitem(1.0))
        assert_q1_equal(gpu, ref)
        # the public Q1 answer pins (combo 0 = A|F)
        assert gpu.sum_qty[0] == 37734107.0 and gpu.count[0] == 1478493
        assert gpu.count[3] == 2920374   # N|O
        sess.tpch_lineitem_free(cols)

    @pytest.mark.parametrize("count", [1, 2, 3, 63, 64, 1000])
    def test_tiny_sizes(self, sess, count):
        """odd n, sub-wave n, tail handling"""
        cols = sess.tpch_lineitem(1.0, 1, count)
        gpu = sess.q1(cols)
        ref, _ = oracle.q1_exact(oracle.gen_lineitem(1.0, 1, count))
        assert_q1_equal(gpu, ref)
        sess.tpch_lineitem_free(cols)

    def test_cutoff_none_selected(self, sess):
        cols = sess.tpch_lineitem(0.01, 1, 500)
        gpu = sess.q1(cols, cutoff=0)
        assert sum(gpu.count) == 0 and sum(gpu.sum_base) == 0.0
        sess.tpch_lineitem_free(cols)

    def test_cutoff_all_selected(self, sess):
        cols = sess.tpch_lineitem(0.01, 1, 500)
        gpu = sess.q1(cols, cutoff=1 << 30)
        ref, _ = oracle.q1_exact(oracle.gen_lineitem(0.01, 1, 500), cutoff=1 << 30)
        assert sum(gpu.count) == cols.row_count
        assert_q1_equal(gpu, ref)
        sess.tpch_lineitem_free(cols)


class TestQ1ParityMode:
    """Sequential-order kernel == reference naive accumulation, bit-for-bit
    (the operator-level FP parity bar at page sizes)."""

    @pytest.mark.parametrize("count", [100, 2000])
    def test_naive_bit_exact(self, sess, count):
        cols = sess.tpch_lineitem(1.0, 1, count)
        gpu = sess.q1_naive(cols)
        ref = oracle.q1_naive(oracle.gen_lineitem(1.0, 1, count))
        assert_q1_equal(gpu, ref)
        sess.tpch_lineitem_free(cols)
