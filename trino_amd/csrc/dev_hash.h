/* dev_hash.h — device-side restatement of the reference's hash functions
 * (same constants as oracle/hashes.c; reference cites there):
 *  - bigint/date HASH_CODE xxmix: spi/type/AbstractLongType.java:121-125
 *  - DOUBLE -0.0 normalization: spi/type/DoubleType.java:199-215
 *  - combine 31*h, INITIAL_HASH_VALUE=0, NULL_HASH_CODE=0:
 *    CombineHashFunction.java:29-32, HashGenerator.java:20
 *  - murmur3 finalizer: BigintGroupByHash.java:297-300, PagesHash.java:35-51
 *  - partition reductions: HashGenerator.java:41-46 (remote),
 *    LocalPartitionGenerator.java:76-80 (local, XxHash64(reverse))
 */
#pragma once
#include "operators.h"

__device__ __host__ static inline uint64_t d_rotl64(uint64_t x, int r)
{
    return (x << r) | (x >> (64 - r));
}

__device__ static inline uint64_t d_bigint_hash(int64_t v)
{
    return d_rotl64((uint64_t)v * 0xC2B2AE3D27D4EB4FULL, 31) * 0x9E3779B185EBCA87ULL;
}

__device__ static inline uint64_t d_canonical_f64_bits(double v)
{
    if (v == 0.0) v = 0.0;                       /* -0.0 -> +0.0 */
    unsigned long long bits = __double_as_longlong(v);
    if (v != v) bits = 0x7FF8000000000000ULL;    /* canonical NaN for grouping */
    return bits;
}

__device__ static inline uint64_t d_double_hash(double v)
{
    return d_bigint_hash((int64_t)d_canonical_f64_bits(v));   /* doubleToLongBits:
        -0.0 -> +0.0, every NaN -> 0x7FF8000000000000 */
}

__device__ static inline uint64_t d_murmur3_mix(uint64_t h)
{
    h ^= h >> 33; h *= 0xFF51AFD7ED558CCDULL;
    h ^= h >> 33; h *= 0xC4CEB9FE1A85EC53ULL;
    h ^= h >> 33;
    return h;
}

/* canonical typed channel value widened to 64 bits (for key stores) */
struct KColH { const void* data; const uint64_t* valid; int32_t type; int32_t _pad;
               const int32_t* offsets; /* VARCHAR */ };

/* standard XxHash64 over bytes, seed 0 (airlift XxHash64; the VARCHAR
 * HASH_CODE operator, AbstractVariableWidthType) */
__device__ static inline uint64_t d_xxhash64_bytes(const uint8_t* p, int64_t len);

__device__ static inline bool kcol_is_null(const KColH& c, int64_t i)
{
    return c.valid && !((c.valid[i >> 6] >> (i & 63)) & 1);
}

__device__ static inline uint64_t kcol_word(const KColH& c, int64_t i)
{
    switch (c.type) {
        case TG_BIGINT: return (uint64_t)((const int64_t*)c.data)[i];
        case TG_INTEGER: case TG_DATE: return (uint64_t)(int64_t)((const int32_t*)c.data)[i];
        case TG_SMALLINT: return (uint64_t)(int64_t)((const int16_t*)c.data)[i];
        case TG_TINYINT: case TG_BOOLEAN: return (uint64_t)(int64_t)((const int8_t*)c.data)[i];
        default: return d_canonical_f64_bits(((const double*)c.data)[i]);
    }
}

/* signed integer view of a fixed-width channel (masked-aggregate gates) */
__device__ static inline int64_t kcol_sval(const KColH& c, int64_t i)
{
    return (int64_t)kcol_word(c, i);
}

/* per-channel HASH_CODE of the canonical word (types map as the reference's
 * long-based HASH_CODE operators; DOUBLE uses raw (normalized) bits) */
__device__ static inline uint64_t kcol_hash(const KColH& c, int64_t i)
{
    if (kcol_is_null(c, i)) return 0;            /* NULL_HASH_CODE */
    if (c.type == TG_DOUBLE) return d_double_hash(((const double*)c.data)[i]);
    if (c.type == TG_VARCHAR)
        return d_xxhash64_bytes((const uint8_t*)c.data + c.offsets[i],
                                c.offsets[i + 1] - c.offsets[i]);
    return d_bigint_hash((int64_t)kcol_word(c, i));
}

__device__ static inline uint64_t row_hash(const KColH* cols, int n, int64_t i)
{
    int64_t h = 0;
    for (int c = 0; c < n; c++)
        h = 31 * h + (int64_t)kcol_hash(cols[c], i);
    return (uint64_t)h;
}

/* HashGenerator.processRawHash: remote hash-distribution partition */
__device__ static inline int32_t d_partition_remote(uint64_t raw, int32_t n)
{
    uint32_t x = (uint32_t)(raw ^ (raw >> 32));   /* Long.hashCode */
    return (int32_t)(((uint64_t)x * (uint64_t)n) >> 32);
}

/* airlift XxHash64.hash(long) (for the local-exchange partition variant) */
__device__ static inline uint64_t d_xxhash64_long(int64_t v)
{
    const uint64_t P1 = 0x9E3779B185EBCA87ULL, P2 = 0xC2B2AE3D27D4EB4FULL,
                   P3 = 0x165667B19E3779F9ULL, P4 = 0x85EBCA77C2B2AE63ULL,
                   P5 = 0x27D4EB2F165667C5ULL;
    uint64_t h = P5 + 8;
    uint64_t k = d_rotl64((uint64_t)v * P2, 31) * P1;
    h = d_rotl64(h ^ k, 27) * P1 + P4;
    h ^= h >> 33; h *= P2; h ^= h >> 29; h *= P3; h ^= h >> 32;
    return h;
}

__device__ inline uint64_t d_xxhash64_bytes(const uint8_t* p, int64_t len)
{
    const uint64_t P1 = 0x9E3779B185EBCA87ULL, P2 = 0xC2B2AE3D27D4EB4FULL,
                   P3 = 0x165667B19E3779F9ULL, P4 = 0x85EBCA77C2B2AE63ULL,
                   P5 = 0x27D4EB2F165667C5ULL;
    const uint8_t* end = p + len;
    uint64_t h;
    auto rd64 = [](const uint8_t* q) {
        uint64_t v;
        memcpy(&v, q, 8);
        return v;
    };
    auto mix = [&](uint64_t cur, uint64_t v) { return d_rotl64(cur + v * P2, 31) * P1; };
    if (len >= 32) {
        uint64_t v1 = P1 + P2, v2 = P2, v3 = 0, v4 = (uint64_t)0 - P1;
        do {
            v1 = mix(v1, rd64(p));
            v2 = mix(v2, rd64(p + 8));
            v3 = mix(v3, rd64(p + 16));
            v4 = mix(v4, rd64(p + 24));
            p += 32;
        } while (p <= end - 32);
        h = d_rotl64(v1, 1) + d_rotl64(v2, 7) + d_rotl64(v3, 12) + d_rotl64(v4, 18);
        h = (h ^ mix(0, v1)) * P1 + P4;
        h = (h ^ mix(0, v2)) * P1 + P4;
        h = (h ^ mix(0, v3)) * P1 + P4;
        h = (h ^ mix(0, v4)) * P1 + P4;
    }
    else {
        h = P5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) {
        h = d_rotl64(h ^ mix(0, rd64(p)), 27) * P1 + P4;
        p += 8;
    }
    if (p + 4 <= end) {
        uint32_t k;
        memcpy(&k, p, 4);
        h = d_rotl64(h ^ ((uint64_t)k * P1), 23) * P2 + P3;
        p += 4;
    }
    while (p < end) {
        h = d_rotl64(h ^ (*p * P5), 11) * P1;
        p++;
    }
    h ^= h >> 33; h *= P2; h ^= h >> 29; h *= P3; h ^= h >> 32;
    return h;
}

__device__ static inline int32_t d_partition_local(uint64_t raw, int32_t n_pow2)
{
    return (int32_t)(d_xxhash64_long((int64_t)__brevll(raw))) & (n_pow2 - 1);
}

/* build device KColH array from a DevPage (subset of channels; caller frees) */
static inline tg_status make_kcols(tg_session* s, const DevPage& page,
                                   const int32_t* channels, int n, KColH** out)
{
    std::vector<KColH> h(n);
    for (int i = 0; i < n; i++) {
        const DevBlock& b = page.blocks[channels ? channels[i] : i];
        h[i] = {b.data, b.valid, (int32_t)b.type, 0, b.offsets};
    }
    TG_HIP_CHECK(hipMalloc(out, n * sizeof(KColH)));
    TG_HIP_CHECK(hipMemcpyAsync(*out, h.data(), n * sizeof(KColH),
                                hipMemcpyHostToDevice, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}
