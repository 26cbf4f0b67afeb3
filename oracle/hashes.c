/* hashes.c — CPU oracle restatement of the reference's hash functions.
 * ORACLE / TEST INFRASTRUCTURE ONLY (see tpch_core.h banner).
 *
 * Restated from:
 *  - bigint/date HASH_CODE ("xxHash64 mix"):
 *      core/trino-spi/.../spi/type/AbstractLongType.java:121-125
 *      rotateLeft(v * 0xC2B2AE3D27D4EB4F, 31) * 0x9E3779B185EBCA87
 *  - DOUBLE HASH_CODE normalizes -0.0 then hashes raw bits:
 *      core/trino-spi/.../spi/type/DoubleType.java:199-215
 *  - murmur3 64-bit finalizer (fastutil HashCommon.murmurHash3; used by
 *      operator/BigintGroupByHash.java:297-300 and operator/join/PagesHash.java:35-51,
 *      constants 0xFF51AFD7ED558CCD / 0xC4CEB9FE1A85EC53)
 *  - combine: 31*prev + h:
 *      core/trino-main/.../operator/scalar/CombineHashFunction.java:29-32,
 *      HashGenerator.java:20 (INITIAL_HASH_VALUE=0), TypeUtils NULL_HASH_CODE=0
 *  - VARCHAR HASH_CODE = XxHash64 over the utf8 bytes, seed 0 (airlift
 *      XxHash64; spi AbstractVariableWidthType:368-378). XxHash64 is the
 *      standard public xxHash64 algorithm; pinned by its public test vectors.
 *  - local exchange partition: (int) XxHash64.hash(Long.reverse(rawHash)) & mask
 *      operator/exchange/LocalPartitionGenerator.java:76-80
 *  - remote partition: (unsigned(Long.hashCode(h)) * partitionCount) >>> 32
 *      operator/HashGenerator.java:41-46
 */
#include <stdint.h>
#include <stddef.h>
#include <string.h>

#define EXPORT __attribute__((visibility("default")))

static inline uint64_t rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }

/* AbstractLongType.hash */
EXPORT uint64_t o_bigint_hash(int64_t v)
{
    return rotl64((uint64_t)v * 0xC2B2AE3D27D4EB4FULL, 31) * 0x9E3779B185EBCA87ULL;
}

/* DoubleType hashCodeOperator */
EXPORT uint64_t o_double_hash(double v)
{
    if (v == 0) v = 0;          /* normalizes -0.0 to +0.0 */
    uint64_t bits;
    memcpy(&bits, &v, 8);
    /* Java doubleToLongBits canonicalizes every NaN to 0x7FF8000000000000 */
    if (v != v) bits = 0x7FF8000000000000ULL;
    return o_bigint_hash((int64_t)bits);
}

/* fastutil HashCommon.murmurHash3 (64-bit finalizer) */
EXPORT uint64_t o_murmur3_mix(uint64_t h)
{
    h ^= h >> 33;
    h *= 0xFF51AFD7ED558CCDULL;
    h ^= h >> 33;
    h *= 0xC4CEB9FE1A85EC53ULL;
    h ^= h >> 33;
    return h;
}

EXPORT int64_t o_combine_hash(int64_t prev, int64_t h) { return 31 * prev + h; }

/* ---- XxHash64, standard algorithm, seed 0 (airlift io.airlift.slice.XxHash64) */
#define P1 0x9E3779B185EBCA87ULL
#define P2 0xC2B2AE3D27D4EB4FULL
#define P3 0x165667B19E3779F9ULL
#define P4 0x85EBCA77C2B2AE63ULL
#define P5 0x27D4EB2F165667C5ULL

static inline uint64_t xx_mix(uint64_t cur, uint64_t v)
{
    return rotl64(cur + v * P2, 31) * P1;
}
static inline uint64_t xx_final(uint64_t h)
{
    h ^= h >> 33; h *= P2; h ^= h >> 29; h *= P3; h ^= h >> 32;
    return h;
}

EXPORT uint64_t o_xxhash64(const void* data, size_t len)
{
    const uint8_t* p = (const uint8_t*)data;
    const uint8_t* end = p + len;
    uint64_t h;
    if (len >= 32) {
        uint64_t v1 = 0 + P1 + P2, v2 = P2, v3 = 0, v4 = 0 - P1;
        do {
            uint64_t k;
            memcpy(&k, p, 8);      v1 = xx_mix(v1, k);
            memcpy(&k, p + 8, 8);  v2 = xx_mix(v2, k);
            memcpy(&k, p + 16, 8); v3 = xx_mix(v3, k);
            memcpy(&k, p + 24, 8); v4 = xx_mix(v4, k);
            p += 32;
        } while (p <= end - 32);
        h = rotl64(v1, 1) + rotl64(v2, 7) + rotl64(v3, 12) + rotl64(v4, 18);
        h = (h ^ xx_mix(0, v1)) * P1 + P4;
        h = (h ^ xx_mix(0, v2)) * P1 + P4;
        h = (h ^ xx_mix(0, v3)) * P1 + P4;
        h = (h ^ xx_mix(0, v4)) * P1 + P4;
    }
    else {
        h = P5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) {
        uint64_t k; memcpy(&k, p, 8);
        h = rotl64(h ^ xx_mix(0, k), 27) * P1 + P4;
        p += 8;
    }
    if (p + 4 <= end) {
        uint32_t k; memcpy(&k, p, 4);
        h = rotl64(h ^ ((uint64_t)k * P1), 23) * P2 + P3;
        p += 4;
    }
    while (p < end) {
        h = rotl64(h ^ (*p * P5), 11) * P1;
        p++;
    }
    return xx_final(h);
}

/* airlift XxHash64.hash(long) — xxh64 of the 8-byte LE value, seed 0 */
EXPORT uint64_t o_xxhash64_long(int64_t v)
{
    uint64_t h = P5 + 8;
    h = rotl64(h ^ xx_mix(0, (uint64_t)v), 27) * P1 + P4;
    return xx_final(h);
}

static inline uint64_t bit_reverse64(uint64_t x)
{
    x = ((x & 0x5555555555555555ULL) << 1)  | ((x >> 1)  & 0x5555555555555555ULL);
    x = ((x & 0x3333333333333333ULL) << 2)  | ((x >> 2)  & 0x3333333333333333ULL);
    x = ((x & 0x0F0F0F0F0F0F0F0FULL) << 4)  | ((x >> 4)  & 0x0F0F0F0F0F0F0F0FULL);
    x = ((x & 0x00FF00FF00FF00FFULL) << 8)  | ((x >> 8)  & 0x00FF00FF00FF00FFULL);
    x = ((x & 0x0000FFFF0000FFFFULL) << 16) | ((x >> 16) & 0x0000FFFF0000FFFFULL);
    x = (x << 32) | (x >> 32);
    return x;
}

/* LocalPartitionGenerator.processRawHash & hashMask (partitionCount pow2) */
EXPORT int32_t o_partition_local(int64_t raw_hash, int32_t partition_count)
{
    return (int32_t)o_xxhash64_long((int64_t)bit_reverse64((uint64_t)raw_hash))
           & (partition_count - 1);
}

/* HashGenerator.processRawHash (remote hash distribution) */
EXPORT int32_t o_partition_remote(int64_t raw_hash, int32_t partition_count)
{
    uint32_t x = (uint32_t)((uint64_t)raw_hash ^ ((uint64_t)raw_hash >> 32)); /* Long.hashCode */
    return (int32_t)(((uint64_t)x * (uint64_t)partition_count) >> 32);
}

/* ---- canonical row hash over typed channels ----
 * h=0; per channel h = 31*h + (null ? 0 : typeHash(value))
 * types: 0=BIGINT 1=INTEGER(date) 4=DOUBLE 5=DATE 7=VARCHAR(offsets+bytes) 3=TINYINT
 * (matches include/trino_gpu.h tg_type; INTEGER/DATE/TINYINT hash as their
 *  Java long value through AbstractLongType-style hash of the widened value —
 *  IntegerType/DateType HASH_CODE use the same xxmix of the long value.) */
EXPORT void o_hash_rows(int32_t n_channels, const int32_t* types,
                        const void* const* datas, const int32_t* const* offsets,
                        int64_t n, uint64_t* out)
{
    for (int64_t i = 0; i < n; i++) out[i] = 0;
    for (int32_t c = 0; c < n_channels; c++) {
        int t = types[c];
        for (int64_t i = 0; i < n; i++) {
            uint64_t h;
            switch (t) {
                case 0: h = o_bigint_hash(((const int64_t*)datas[c])[i]); break;
                case 1: case 5: h = o_bigint_hash(((const int32_t*)datas[c])[i]); break;
                case 2: h = o_bigint_hash(((const int16_t*)datas[c])[i]); break;
                case 3: h = o_bigint_hash(((const int8_t*)datas[c])[i]); break;
                case 4: h = o_double_hash(((const double*)datas[c])[i]); break;
                case 7: {
                    const int32_t* off = offsets[c];
                    h = o_xxhash64((const uint8_t*)datas[c] + off[i], (size_t)(off[i+1] - off[i]));
                    break;
                }
                default: h = 0;
            }
            out[i] = (uint64_t)(31 * (int64_t)out[i] + (int64_t)h);
        }
    }
}
