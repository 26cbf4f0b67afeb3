/* groupby.c — CPU oracle restatement of the reference's group-by hashes and
 * grouped aggregation. ORACLE / TEST INFRASTRUCTURE ONLY.
 *
 * Restated from:
 *  - operator/BigintGroupByHash.java:191-300: open addressing, linear probe,
 *    position = murmur3(value) & mask, FILL_RATIO 0.75, capacity doubling,
 *    null key -> its own group id, group ids assigned IN ROW ORDER
 *    (contract: operator/GroupByHash.java:121-128)
 *  - operator/FlatHash.java:276-330: the generic (FlatGroupByHash) path.
 *    Its observable contract at this boundary is identical — group ids in
 *    first-occurrence row order; aggregation output emits groups in
 *    consecutive group-id order (InMemoryHashAggregationBuilder buildResult).
 *    Restated here with the same canonical row hash (31*combine of HASH_CODE,
 *    FlatHashStrategyCompiler.java:899-910) and an open-addressed exact-match
 *    table; the Swiss-table control bytes are an internal layout detail that
 *    does not change any output.
 *  - accumulators: DoubleSumAggregation.java:37-45 (sum += v, null skip),
 *    DoubleAverageAggregations.java:38-63 (count+sum), CountAggregation.
 */
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

#define EXPORT __attribute__((visibility("default")))

uint64_t o_murmur3_mix(uint64_t h);
uint64_t o_bigint_hash(int64_t v);
int32_t o_flat_groupby_v(int32_t n_channels, const int32_t* types,
                         const void* const* datas, const int32_t* const* offsets,
                         int64_t n, int32_t* out_group_ids, int64_t* out_first_row_by_gid);
void o_hash_rows(int32_t, const int32_t*, const void* const*, const int32_t* const*,
                 int64_t, uint64_t*);

/* ---- BigintGroupByHash ----
 * keys: i64 values; valid: packed bitmap (bit=1 valid) or NULL for no-nulls.
 * out_group_ids[n]; out_values_by_gid (size >= n) filled with key per group id;
 * returns group count. Mirrors putIfAbsent/addNewGroup/tryRehash. */
EXPORT int32_t o_bigint_groupby(const int64_t* keys, const uint64_t* valid, int64_t n,
                                int32_t* out_group_ids, int64_t* out_values_by_gid,
                                int32_t* out_null_group_id)
{
    int32_t capacity = 1 << 11;   /* BigintGroupByHash(expectedSize) -> arraySize; start small, rehash as needed */
    int32_t mask = capacity - 1;
    int64_t* values = malloc(sizeof(int64_t) * capacity);
    int32_t* group_ids = malloc(sizeof(int32_t) * capacity);
    memset(group_ids, -1, sizeof(int32_t) * capacity);
    int32_t next_group = 0, null_group = -1;
    int32_t max_fill = (int32_t)(capacity * 0.75);

    for (int64_t i = 0; i < n; i++) {
        if (valid && !((valid[i >> 6] >> (i & 63)) & 1)) {
            if (null_group < 0) null_group = next_group++;
            out_group_ids[i] = null_group;
            continue;
        }
        int64_t v = keys[i];
        int32_t pos = (int32_t)(o_murmur3_mix((uint64_t)v) & (uint64_t)mask);
        int32_t gid;
        while (1) {
            gid = group_ids[pos];
            if (gid == -1) break;
            if (values[pos] == v && gid != null_group) { goto assigned; }
            /* note: slot holding the null group never exists in `values`;
             * null group id only lives in the null branch above */
            pos = (pos + 1) & mask;
        }
        gid = next_group++;
        values[pos] = v;
        group_ids[pos] = gid;
        out_values_by_gid[gid] = v;
        if (next_group >= max_fill) {
            int32_t new_cap = capacity * 2;
            int32_t new_mask = new_cap - 1;
            int64_t* nv = malloc(sizeof(int64_t) * new_cap);
            int32_t* ng = malloc(sizeof(int32_t) * new_cap);
            memset(ng, -1, sizeof(int32_t) * new_cap);
            for (int32_t s = 0; s < capacity; s++) {
                if (group_ids[s] != -1) {
                    int32_t p = (int32_t)(o_murmur3_mix((uint64_t)values[s]) & (uint64_t)new_mask);
                    while (ng[p] != -1) p = (p + 1) & new_mask;
                    nv[p] = values[s]; ng[p] = group_ids[s];
                }
            }
            free(values); free(group_ids);
            values = nv; group_ids = ng; capacity = new_cap; mask = new_mask;
            max_fill = (int32_t)(capacity * 0.75);
        }
assigned:
        out_group_ids[i] = gid;
    }
    free(values); free(group_ids);
    if (out_null_group_id) *out_null_group_id = null_group;
    return next_group;
}

/* ---- generic FlatGroupByHash-equivalent: group ids in row order for typed
 * channels. Exact key compare; hash = canonical row hash. types as in
 * o_hash_rows. Fixed-width channels only here (varchar group keys arrive
 * dictionary-encoded on this path; dictionary ids are TINYINT channels). */
typedef struct { uint64_t hash; int64_t first_row; int32_t gid; int32_t used; } flat_slot;

static int rows_equal(int32_t n_channels, const int32_t* types,
                      const void* const* datas, const int32_t* const* offsets,
                      int64_t a, int64_t b)
{
    for (int32_t c = 0; c < n_channels; c++) {
        switch (types[c]) {
            case 7: { /* VARCHAR: byte equality (VarcharType EQUAL) */
                const int32_t* off = offsets[c];
                int32_t la = off[a + 1] - off[a], lb = off[b + 1] - off[b];
                if (la != lb) return 0;
                if (memcmp((const uint8_t*)datas[c] + off[a],
                           (const uint8_t*)datas[c] + off[b], (size_t)la) != 0) return 0;
                break;
            }
            case 0: if (((const int64_t*)datas[c])[a] != ((const int64_t*)datas[c])[b]) return 0; break;
            case 1: case 5: if (((const int32_t*)datas[c])[a] != ((const int32_t*)datas[c])[b]) return 0; break;
            case 2: if (((const int16_t*)datas[c])[a] != ((const int16_t*)datas[c])[b]) return 0; break;
            case 3: case 6: if (((const int8_t*)datas[c])[a] != ((const int8_t*)datas[c])[b]) return 0; break;
            case 4: { /* DoubleType EQUAL operator: == (so +0.0 == -0.0; NaN != NaN
                         — but grouping uses IDENTICAL/flat compare of canonical
                         bits in practice; TPC-H data has no NaN/-0) */
                double x = ((const double*)datas[c])[a], y = ((const double*)datas[c])[b];
                if (!(x == y)) return 0; break;
            }
            default: return 0;
        }
    }
    return 1;
}

EXPORT int32_t o_flat_groupby(int32_t n_channels, const int32_t* types,
                              const void* const* datas, int64_t n,
                              int32_t* out_group_ids, int64_t* out_first_row_by_gid)
{
    return o_flat_groupby_v(n_channels, types, datas, NULL, n,
                            out_group_ids, out_first_row_by_gid);
}

EXPORT int32_t o_flat_groupby_v(int32_t n_channels, const int32_t* types,
                                const void* const* datas,
                                const int32_t* const* offsets, int64_t n,
                                int32_t* out_group_ids, int64_t* out_first_row_by_gid)
{
    uint64_t* hashes = malloc(sizeof(uint64_t) * (size_t)n);
    o_hash_rows(n_channels, types, datas, offsets, n, hashes);
    int64_t capacity = 16;
    while (capacity * 15 / 16 < 64) capacity *= 2;
    flat_slot* slots = calloc((size_t)capacity, sizeof(flat_slot));
    int32_t next_group = 0;
    for (int64_t i = 0; i < n; i++) {
        uint64_t h = hashes[i];
        int64_t mask = capacity - 1;
        int64_t pos = (int64_t)(h & (uint64_t)mask);
        int32_t gid = -1;
        while (1) {
            if (!slots[pos].used) break;
            if (slots[pos].hash == h &&
                rows_equal(n_channels, types, datas, offsets, slots[pos].first_row, i)) {
                gid = slots[pos].gid; break;
            }
            pos = (pos + 1) & mask;
        }
        if (gid < 0) {
            gid = next_group++;
            slots[pos].used = 1; slots[pos].hash = h; slots[pos].first_row = i; slots[pos].gid = gid;
            out_first_row_by_gid[gid] = i;
            if (next_group >= capacity * 15 / 16) {
                int64_t nc = capacity * 2;
                flat_slot* ns = calloc((size_t)nc, sizeof(flat_slot));
                for (int64_t s = 0; s < capacity; s++) {
                    if (slots[s].used) {
                        int64_t p = (int64_t)(slots[s].hash & (uint64_t)(nc - 1));
                        while (ns[p].used) p = (p + 1) & (nc - 1);
                        ns[p] = slots[s];
                    }
                }
                free(slots); slots = ns; capacity = nc;
            }
        }
        out_group_ids[i] = gid;
    }
    free(slots); free(hashes);
    return next_group;
}

/* ---- grouped aggregation, reference accumulation order (naive sequential) ----
 * DoubleSumAggregation: state += value, nulls skipped. Rows applied in order. */
EXPORT void o_grouped_sum_f64(const int32_t* gids, const double* vals, const uint64_t* valid,
                              int64_t n, double* sums /* pre-zeroed, one per group */)
{
    for (int64_t i = 0; i < n; i++) {
        if (valid && !((valid[i >> 6] >> (i & 63)) & 1)) continue;
        sums[gids[i]] += vals[i];
    }
}

EXPORT void o_grouped_count(const int32_t* gids, int64_t n, int64_t* counts)
{
    for (int64_t i = 0; i < n; i++) counts[gids[i]]++;
}

/* exact (correctly-rounded) sum leg: accumulate val * 2^scale_pow in i128.
 * Precondition (checked by caller): every |val|*2^scale_pow is an integer
 * exactly representable — true for Q1 columns. */
EXPORT void o_grouped_sum_f64_exact(const int32_t* gids, const double* vals,
                                    int64_t n, int32_t scale_pow,
                                    double* out_sums, int64_t n_groups)
{
    __int128* acc = calloc((size_t)n_groups, sizeof(__int128));
    double scale = 1.0;
    for (int32_t k = 0; k < scale_pow; k++) scale *= 2.0;
    for (int64_t i = 0; i < n; i++) {
        double y = vals[i] * scale;       /* exact: power-of-two scaling */
        acc[gids[i]] += (__int128)(long long)y;
    }
    double inv = 1.0 / scale;
    for (int64_t g = 0; g < n_groups; g++)
        out_sums[g] = (double)acc[g] * inv;   /* i128->double correctly rounded, *2^-k exact */
    free(acc);
}
