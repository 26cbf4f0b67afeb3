"""Oracle unit pins: hashes, group-by contract, join chain order, partition.

Golden sources: public xxHash64 test vectors; the reference's constants
(AbstractLongType.java:121-125, CombineHashFunction.java:29-32,
BigintGroupByHash.java:297-300, GroupByHash.java:121-128 row-order contract,
ArrayPositionLinks.java:24-45 chain semantics, HashGenerator.java:41-46 and
LocalPartitionGenerator.java:76-80 partition reduction). Python reimplements
each formula independently to cross-check the C restatement.
"""
import numpy as np
import pytest

import oracle

M64 = (1 << 64) - 1


def rotl(x, r):
    return ((x << r) | (x >> (64 - r))) & M64


def py_bigint_hash(v):
    return (rotl((v & M64) * 0xC2B2AE3D27D4EB4F & M64, 31) * 0x9E3779B185EBCA87) & M64


def py_murmur3(h):
    h &= M64
    h ^= h >> 33
    h = (h * 0xFF51AFD7ED558CCD) & M64
    h ^= h >> 33
    h = (h * 0xC4CEB9FE1A85EC53) & M64
    h ^= h >> 33
    return h


def test_bigint_hash():
    for v in [0, 1, -1, 42, 2**62, -2**62, 123456789123456789]:
        assert oracle.bigint_hash(v) == py_bigint_hash(v)


def test_murmur3_mix():
    for v in [0, 1, 0xDEADBEEF, 2**63, M64]:
        assert oracle.murmur3_mix(v) == py_murmur3(v)


def test_combine_hash():
    assert oracle.combine_hash(0, 5) == 5
    assert oracle.combine_hash(7, 3) == 31 * 7 + 3
    # overflow wraps like Java long
    assert oracle.combine_hash(2**62, 2**62) == ((31 * 2**62 + 2**62 + 2**63) % 2**64) - 2**63


def test_double_hash_normalizes_negative_zero():
    assert oracle.double_hash(-0.0) == oracle.double_hash(0.0)
    assert oracle.double_hash(1.5) == py_bigint_hash(np.float64(1.5).view(np.int64).item())


def test_xxhash64_public_vectors():
    # public xxHash64 reference vectors, seed 0
    assert oracle.xxhash64(b"") == 0xEF46DB3751D8E999
    assert oracle.xxhash64(b"a") == 0xD24EC4F1A98C6E5B
    assert oracle.xxhash64(b"abc") == 0x44BC2CF5AD770999
    assert oracle.xxhash64(b"as") == 0x1C330FB2D66BE179
    assert oracle.xxhash64(b"asd") == 0x631C37CE72A97393
    assert oracle.xxhash64(b"asdf") == 0x415872F599CEA71E
    # >32 bytes path
    data = bytes(range(64))
    assert oracle.xxhash64(data) == oracle.xxhash64(data)  # deterministic
    # long-value specialization == bytes of LE long
    for v in [0, 1, -1, 123456789]:
        assert oracle.xxhash64_long(v) == oracle.xxhash64(
            int(v).to_bytes(8, "little", signed=True))


def test_partition_functions():
    for h in [0, 1, -5, 2**40, -2**40]:
        # local: (int)XxHash64.hash(Long.reverse(h)) & mask
        rev = int(bin(h & M64)[2:].zfill(64)[::-1], 2)
        x = oracle.xxhash64_long(rev - 2**64 if rev >= 2**63 else rev)
        assert oracle.partition_local(h, 8) == (x & 0xFFFFFFFF) % 2**32 & 7
        # remote: (unsigned(Long.hashCode(h)) * n) >>> 32
        lh = ((h & M64) ^ ((h & M64) >> 32)) & 0xFFFFFFFF
        assert oracle.partition_remote(h, 13) == (lh * 13) >> 32


def test_hash_rows_combine():
    k1 = np.array([1, 2, 3], np.int64)
    k2 = np.array([4.0, -0.0, 1.5], np.float64)
    h = oracle.hash_rows([k1, k2], [oracle.TG_BIGINT, oracle.TG_DOUBLE])
    for i in range(3):
        exp = (31 * py_bigint_hash(int(k1[i])) +
               py_bigint_hash(np.float64(abs(k2[i]) if k2[i] == 0 else k2[i]).view(np.int64).item())) & M64
        assert h[i] == exp


class TestBigintGroupBy:
    """Contract of GroupByHash.getGroupIds (GroupByHash.java:121-128):
    ids assigned in row order of first occurrence; null gets its own id."""

    def test_row_order_ids(self):
        keys = np.array([9, 9, 3, 9, 5, 3, 7], np.int64)
        gids, ng, vals, nullg = oracle.bigint_groupby(keys)
        assert list(gids) == [0, 0, 1, 0, 2, 1, 3]
        assert ng == 4 and list(vals) == [9, 3, 5, 7] and nullg == -1

    def test_null_group(self):
        keys = np.array([9, 0, 3, 0], np.int64)
        valid = np.array([0b0101], np.uint64)  # rows 1,3 null
        gids, ng, vals, nullg = oracle.bigint_groupby(keys, valid)
        assert list(gids) == [0, 1, 2, 1]
        assert ng == 3 and nullg == 1

    def test_rehash_preserves_ids(self):
        n = 100_000
        rng = np.random.default_rng(0)
        keys = rng.integers(0, 5000, n).astype(np.int64)
        gids, ng, vals, _ = oracle.bigint_groupby(keys)
        # ids must match first-occurrence order
        seen = {}
        exp = np.empty(n, np.int32)
        for i, k in enumerate(keys.tolist()):
            exp[i] = seen.setdefault(k, len(seen))
        assert np.array_equal(gids, exp) and ng == len(seen)
        assert np.array_equal(vals, np.array(list(seen.keys()), np.int64))


class TestFlatGroupBy:
    def test_multi_channel(self):
        c1 = np.array([1, 1, 2, 1, 2], np.int64)
        c2 = np.array([1.0, 2.0, 1.0, 1.0, 1.0], np.float64)
        gids, ng, first = oracle.flat_groupby([c1, c2], [oracle.TG_BIGINT, oracle.TG_DOUBLE])
        assert list(gids) == [0, 1, 2, 0, 2] and ng == 3
        assert list(first) == [0, 1, 2]

    def test_many_groups_rehash(self):
        n = 50_000
        rng = np.random.default_rng(1)
        c1 = rng.integers(0, 300, n).astype(np.int8)
        c2 = rng.integers(0, 50, n).astype(np.int32)
        gids, ng, first = oracle.flat_groupby([c1, c2], [oracle.TG_TINYINT, oracle.TG_INTEGER])
        seen = {}
        exp = np.empty(n, np.int32)
        for i in range(n):
            exp[i] = seen.setdefault((int(c1[i]), int(c2[i])), len(seen))
        assert np.array_equal(gids, exp) and ng == len(seen)


class TestJoin:
    def test_chain_order_reverse_insertion(self):
        """ArrayPositionLinks: duplicate key matches emit newest build row
        first, then links to older rows (DefaultPagesHash.insertValue)."""
        bk = np.array([10, 20, 10, 30, 10], np.int64)
        t = oracle.JoinTable(bk)
        op, ob = t.probe(np.array([10, 25, 30], np.int64))
        assert list(zip(op.tolist(), ob.tolist())) == [(0, 4), (0, 2), (0, 0), (2, 3)]

    def test_null_keys_never_match(self):
        bk = np.array([10, 20], np.int64)
        bvalid = np.array([0b01], np.uint64)  # row1 (20) is null
        t = oracle.JoinTable(bk, valid=bvalid)
        pk = np.array([20, 10], np.int64)
        pvalid = np.array([0b01], np.uint64)  # probe row1 (10) null
        op, ob = t.probe(pk, probe_valid=pvalid)
        assert len(op) == 0

    def test_load_factor_sizing(self):
        # IncrementalLoadFactorHashArraySizeSupplier: <=65536 -> 0.25
        t = oracle.JoinTable(np.arange(100, dtype=np.int64))
        assert t.table_size() == 512  # 100/0.25=400 -> 512

    def test_grouped_aggregation(self):
        gids = np.array([0, 1, 0, 1, 0], np.int32)
        vals = np.array([1.5, 2.0, 2.5, 3.0, -1.0], np.float64)
        s = oracle.grouped_sum_f64(gids, vals, 2)
        assert s[0] == 1.5 + 2.5 + -1.0 and s[1] == 5.0
        c = oracle.grouped_count(gids, 2)
        assert list(c) == [3, 2]
