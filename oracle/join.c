/* join.c — CPU oracle restatement of the reference's hash join build/probe.
 * ORACLE / TEST INFRASTRUCTURE ONLY.
 *
 * Restated from:
 *  - build: operator/PagesIndex.java:91-92,224-253 (synthetic addresses
 *    (sliceIndex<<32)|position, SyntheticAddress.java:26-29),
 *    operator/join/DefaultPagesHash.java:34-153 (power-of-2 table of int
 *    address-indexes, 1-byte hash tags, linear probe; insert links duplicate
 *    keys via ArrayPositionLinks.java:24-45: positionLinks[new]=old and the
 *    NEW row becomes the slot head), BigintPagesHash.java:62-143 (single
 *    BIGINT key specialization), sizing
 *    IncrementalLoadFactorHashArraySizeSupplier.java:26-45 (0.25/0.5/0.75 by
 *    build count), hash position = murmur3 finalizer & mask
 *    (operator/join/PagesHash.java:35-51).
 *  - probe: operator/join/DefaultPagesHash.java:193-280 batched
 *    getAddressIndex (hash-all -> gather-all -> verify -> linear-probe the
 *    misses) and JoinProbe/DefaultPageJoiner chain walk
 *    (join/DefaultPageJoiner.java:243-296): per probe row emit the head
 *    match then follow links (= reverse build insertion order); null keys
 *    never match.
 * Observables at this boundary: the (probe_row, build_address) pair sequence.
 * Chain order per key is reverse insertion order for ANY hash function
 * (equal keys always share a slot), so the restatement is exact.
 */
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

#define EXPORT __attribute__((visibility("default")))

uint64_t o_murmur3_mix(uint64_t h);

typedef struct {
    int64_t n;            /* build rows = address count */
    int32_t* keys_table;  /* power-of-2, -1 empty, else address index */
    int64_t  mask;
    int32_t* links;       /* ArrayPositionLinks: links[addr] = next addr or -1 */
    int64_t* key_vals;    /* build key per address index */
    uint8_t* tags;        /* positionToHashes 1-byte tag */
    int64_t* addresses;   /* synthetic addresses (slice<<32)|pos */
} o_join_table;

/* IncrementalLoadFactorHashArraySizeSupplier (multiplier 1) + HashCommon.arraySize */
static int64_t hash_array_size(int64_t expected)
{
    double f = expected <= (1 << 16) ? 0.25 : expected <= (1 << 20) ? 0.50 : 0.75;
    int64_t need = (int64_t)(expected / f);
    if (need < 2) need = 2;
    int64_t cap = 1;
    while (cap < need) cap <<= 1;
    return cap;
}

/* Build from a BIGINT key column. positions_per_page: rows per PagesIndex
 * slice (addresses encode (slice<<32)|pos); pass n for a single page.
 * valid: packed bitmap (bit=1 valid) or NULL. Null keys are indexed as
 * non-matching (skipped at insert, like isPositionNull -> continue). */
EXPORT o_join_table* o_join_build_bigint(const int64_t* keys, const uint64_t* valid,
                                         int64_t n, int64_t positions_per_page)
{
    o_join_table* t = calloc(1, sizeof(o_join_table));
    t->n = n;
    int64_t cap = hash_array_size(n);
    t->mask = cap - 1;
    t->keys_table = malloc(sizeof(int32_t) * cap);
    memset(t->keys_table, -1, sizeof(int32_t) * cap);
    t->links = malloc(sizeof(int32_t) * n);
    memset(t->links, -1, sizeof(int32_t) * n);
    t->key_vals = malloc(sizeof(int64_t) * n);
    t->tags = malloc(n);
    t->addresses = malloc(sizeof(int64_t) * n);
    for (int64_t i = 0; i < n; i++) {
        t->addresses[i] = ((i / positions_per_page) << 32) | (i % positions_per_page);
        t->key_vals[i] = keys[i];
        uint64_t h = o_murmur3_mix((uint64_t)keys[i]);
        t->tags[i] = (uint8_t)h;
        if (valid && !((valid[i >> 6] >> (i & 63)) & 1)) continue;  /* null: never indexed */
        int64_t pos = (int64_t)(h & (uint64_t)t->mask);
        int32_t insert = (int32_t)i;
        while (t->keys_table[pos] != -1) {
            int32_t cur = t->keys_table[pos];
            if (t->tags[cur] == t->tags[i] && t->key_vals[cur] == keys[i]) {
                /* ArrayPositionLinks.link(new, old): new points at old, new is head */
                t->links[insert] = cur;
                break;
            }
            pos = (pos + 1) & t->mask;
        }
        t->keys_table[pos] = insert;
    }
    return t;
}

EXPORT void o_join_table_free(o_join_table* t)
{
    if (!t) return;
    free(t->keys_table); free(t->links); free(t->key_vals); free(t->tags); free(t->addresses);
    free(t);
}

EXPORT int64_t o_join_table_size(const o_join_table* t) { return t->mask + 1; }

/* getAddressIndex for one probe key: head address index or -1 */
static int32_t probe_one(const o_join_table* t, int64_t key)
{
    uint64_t h = o_murmur3_mix((uint64_t)key);
    int64_t pos = (int64_t)(h & (uint64_t)t->mask);
    while (t->keys_table[pos] != -1) {
        int32_t cur = t->keys_table[pos];
        if (t->tags[cur] == (uint8_t)h && t->key_vals[cur] == key) return cur;
        pos = (pos + 1) & t->mask;
    }
    return -1;
}

/* Inner-join probe of m rows: emits (probe_row, build_address_index) pairs in
 * the reference's order (probe rows ascending; per row, head then links).
 * out arrays sized by caller (cap entries); returns pair count (or -1 if cap
 * exceeded). out_build receives ADDRESS INDEXES; o_join_addresses() maps to
 * synthetic addresses. */
EXPORT int64_t o_join_probe_bigint(const o_join_table* t,
                                   const int64_t* probe_keys, const uint64_t* probe_valid,
                                   int64_t m, int64_t cap,
                                   int32_t* out_probe, int32_t* out_build)
{
    int64_t cnt = 0;
    for (int64_t i = 0; i < m; i++) {
        if (probe_valid && !((probe_valid[i >> 6] >> (i & 63)) & 1)) continue;
        int32_t p = probe_one(t, probe_keys[i]);
        while (p != -1) {
            if (cnt >= cap) return -1;
            out_probe[cnt] = (int32_t)i;
            out_build[cnt] = p;
            cnt++;
            p = t->links[p];
        }
    }
    return cnt;
}

EXPORT const int64_t* o_join_addresses(const o_join_table* t) { return t->addresses; }
