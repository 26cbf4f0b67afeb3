/* ops_join.hip — hash join build + probe (nonspilling).
 *
 * Mirrors:
 *  - build: operator/join/nonspilling/HashBuilderOperator.java:140-380 state
 *    machine (CONSUMING_INPUT -> LOOKUP_SOURCE_BUILT), PagesIndex.java:91-92
 *    page accumulation with synthetic addresses (SyntheticAddress.java:26-29:
 *    (sliceIndex<<32)|position), BigintPagesHash.java:62-143 (single BIGINT
 *    key: values copied to a flat array, open addressing, murmur3 position,
 *    PagesHash.java:35-51), sizing IncrementalLoadFactorHashArraySizeSupplier
 *    .java:26-45 (0.25/0.5/0.75), duplicate keys chained through
 *    ArrayPositionLinks.java:24-45 (links[new]=old, new row becomes head).
 *    The default CSR build sorts each bucket by build row and the probe
 *    walks buckets descending, reproducing the reference's reverse-insertion
 *    duplicate order deterministically at every size. (The legacy
 *    open-addressing path, TG_JOIN_CSR=0, pushes chains with atomicExch and
 *    is only deterministic for single-workgroup builds.)
 *  - probe: LookupJoinOperator / DefaultPageJoiner.java:243-296 +
 *    JoinProbe.java:112-180 batch shape and DefaultPagesHash.java:193-280
 *    (hash all -> gather all -> verify -> probe misses); null keys never
 *    match; per probe row the head match is emitted then the chain is walked.
 *    Output assembly mirrors LookupJoinPageBuilder.java:89-119: probe columns
 *    gathered by probe position, build columns gathered by synthetic address.
 */
#include "dev_hash.h"

/* inline single-entry bucket (the common case: unique build keys): one
 * aligned 16-byte random read resolves most probes instead of three
 * (bucket_off, csr_rows, keys). row -1 = empty bucket, -2 = bucket with
 * >1 rows (fall back to the CSR walk). Built for single-BIGINT-key tables. */
struct SlotKV {
    int64_t key;
    int32_t row;
    int32_t _pad;
};

struct JoinTable {
    int64_t n = 0;               /* build rows */
    int64_t capacity = 0, mask = 0;
    int32_t csr = 0;             /* 1 = CSR bucket index (no probing) */
    int32_t* slots = nullptr;    /* open addressing: -1 empty else row index */
    int32_t* links = nullptr;    /* open addressing: dup chain */
    /* CSR mode: bucket b(=slot) holds rows csr_rows[bucket_off[b] ..
     * bucket_off[b+1]); different keys hashing to one slot share the bucket
     * and are filtered by the key compare (no linear probing). */
    int32_t* bucket_off = nullptr;  /* [capacity+1] */
    int32_t* csr_rows = nullptr;    /* [n] */
    int64_t* csr_keys = nullptr;    /* [n] keys in bucket order (single-key
                                       tables): the probe compare reads one
                                       slot-local line instead of the random
                                       csr_rows -> keys chain */
    int64_t* keys = nullptr;     /* [n] flat copy of build keys */
    uint64_t* key_valid = nullptr; /* packed bitmap or null */
    /* generic multi-channel keys (DefaultPagesHash analog; the fields above
     * are the BigintPagesHash specialization). bkeys points at the bridge's
     * concatenated build channels. */
    int32_t generic = 0;
    int32_t n_key_ch = 1;
    const KColH* bkeys = nullptr;   /* device array[n_key_ch] */
    const SlotKV* kv = nullptr;     /* single-key fast table or null */
    int32_t has_null_key = 0;       /* any build key row null (semi-join 3VL) */
    /* ChannelSet analog (SetBuilderOperator.java): dense-range bitmap for
     * semi-join membership — no positions, no chains */
    const uint64_t* set_bitmap = nullptr;
    int64_t set_min = 0, set_max = 0;
};

/* slot of a build row; generic keys hash with the canonical row hash
 * (InterpretedHashGenerator semantics) finalized by murmur3, like
 * DefaultPagesHash.getHashPosition over strategy raw hashes */
__device__ static inline uint32_t jt_build_slot(const JoinTable& t, int64_t row)
{
    uint64_t h = t.generic ? row_hash(t.bkeys, t.n_key_ch, row)
                           : (uint64_t)t.keys[row];
    return (uint32_t)(d_murmur3_mix(h) & (uint64_t)t.mask);
}

__device__ static inline bool jt_build_null(const JoinTable& t, int64_t row)
{
    if (!t.generic)
        return t.key_valid && !((t.key_valid[row >> 6] >> (row & 63)) & 1);
    for (int c = 0; c < t.n_key_ch; c++)
        if (kcol_is_null(t.bkeys[c], row)) return true;
    return false;
}

/* probe-side context: fast single-bigint pointer OR generic channel array */
struct ProbeKeys {
    const int64_t* pk = nullptr;
    const uint64_t* pvalid = nullptr;
    const KColH* pkg = nullptr;   /* device array[n_key_ch] when generic */
};

__device__ static inline bool probe_null(const JoinTable& t, const ProbeKeys& p, int64_t i)
{
    if (!t.generic)
        return p.pvalid && !((p.pvalid[i >> 6] >> (i & 63)) & 1);
    for (int c = 0; c < t.n_key_ch; c++)
        if (kcol_is_null(p.pkg[c], i)) return true;
    return false;
}

__device__ static inline uint32_t probe_slot(const JoinTable& t, const ProbeKeys& p, int64_t i)
{
    uint64_t h = t.generic ? row_hash(p.pkg, t.n_key_ch, i) : (uint64_t)p.pk[i];
    return (uint32_t)(d_murmur3_mix(h) & (uint64_t)t.mask);
}

__device__ static inline bool probe_matches(const JoinTable& t, const ProbeKeys& p,
                                            int64_t probe_row, int32_t build_row)
{
    if (!t.generic) return t.keys[build_row] == p.pk[probe_row];
    for (int c = 0; c < t.n_key_ch; c++) {
        const KColH& bc = t.bkeys[c];
        const KColH& pc = p.pkg[c];
        if (bc.type == TG_VARCHAR) {   /* VarcharType EQUAL: byte equality */
            int32_t bo = bc.offsets[build_row];
            int32_t bl = bc.offsets[build_row + 1] - bo;
            int32_t po = pc.offsets[probe_row];
            if (bl != pc.offsets[probe_row + 1] - po) return false;
            const uint8_t* bb = (const uint8_t*)bc.data + bo;
            const uint8_t* pb = (const uint8_t*)pc.data + po;
            for (int32_t k = 0; k < bl; k++)
                if (bb[k] != pb[k]) return false;
        }
        else if (kcol_word(bc, build_row) != kcol_word(pc, probe_row))
            return false;
    }
    return true;
}

struct tg_join_bridge {
    tg_session* s = nullptr;
    JoinTable t;
    /* dynamic filter source (DynamicFilterSourceOperator /
     * sql/gen/columnar/DynamicPageFilter.java analog): min/max of the
     * non-null build keys, collected at build finish and pushed into the
     * probe-side scan filter by the pipeline driver */
    int64_t key_min = 0, key_max = 0;
    int64_t key_rows = 0;
    /* concatenated build output channels (contiguous by global build row) */
    std::vector<DevBlock> build_channels;
    std::vector<tg_type> build_types;
    std::vector<int32_t> build_output_channels;
    std::vector<int32_t> key_channels;
    KColH* d_bkeys = nullptr;        /* device array over build key channels */
    bool built = false;
    /* fused dynamic filter (DynamicPageFilter.java analog): when requested
     * BEFORE build finish, a dense-range key membership bitmap is built
     * alongside the index and pushed into probe-side scan kernels. Best
     * effort like the reference: generic keys / huge ranges skip it. */
    bool want_bitmap = false;
};

/* IncrementalLoadFactorHashArraySizeSupplier (multiplier 1) */
static int64_t join_hash_size(int64_t expected)
{
    double f = expected <= (1 << 16) ? 0.25 : expected <= (1 << 20) ? 0.50 : 0.75;
    int64_t need = (int64_t)(expected / f);
    if (need < 2) need = 2;
    int64_t cap = 1;
    while (cap < need) cap <<= 1;
    return cap;
}

__global__ void k_join_init(int32_t* slots, int64_t cap, int32_t* links, int64_t n)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t k = i; k < cap; k += stride) slots[k] = -1;
    for (int64_t k = i; k < n; k += stride) links[k] = -1;
}

__global__ void k_key_minmax(const int64_t* __restrict__ keys,
                             const uint64_t* __restrict__ valid, int64_t n,
                             long long* mn, long long* mx, unsigned long long* cnt)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    long long lmn = INT64_MAX, lmx = INT64_MIN;
    unsigned long long lc = 0;
    for (; i < n; i += stride) {
        if (valid && !((valid[i >> 6] >> (i & 63)) & 1)) continue;
        long long v = keys[i];
        lmn = min(lmn, v); lmx = max(lmx, v); lc++;
    }
    /* block-free: atomics are fine here (once per lane per launch) */
    atomicMin(mn, lmn);
    atomicMax(mx, lmx);
    atomicAdd(cnt, lc);
}

/* ---- CSR build: count -> per-region LDS exclusive scan -> parallel
 * atomic scatter -> per-bucket sort by build row. Replaces the global
 * atomic-CAS insert (measured ~1 G CAS/s = 15 ms per 15M-row build) and
 * the earlier chunk-ordered scatter (serial per chunk with a global-memory
 * cursor RMW per row: 16 ms/step on Q3 SF100). The ascending in-bucket sort
 * plus the descending probe walk reproduces ArrayPositionLinks' reverse-
 * insertion duplicate order (ArrayPositionLinks.java:24-45) DETERMINISTICALLY
 * at every size. ---- */
#define JREG_SLOTS 32768              /* slots per region (LDS u32 counts) */

__global__ void k_jc_count(JoinTable t, uint32_t* __restrict__ slot_of,
                           int32_t* __restrict__ bucket_cnt /* zeroed [cap] */)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < t.n; i += stride) {
        bool null = jt_build_null(t, i);
        uint32_t slot = null ? 0xFFFFFFFFu : jt_build_slot(t, i);
        slot_of[i] = slot;
        if (!null) atomicAdd(&bucket_cnt[slot], 1);
    }
}

__global__ __launch_bounds__(TG_BLOCK)
void k_jc_local_off(const int32_t* __restrict__ bucket_cnt,
                    int32_t* __restrict__ bucket_off,
                    int32_t* __restrict__ region_total,
                    int64_t nreg, int64_t region_slots)
{
    /* one block per region: region-local exclusive prefix of the per-slot
     * counts (LDS staged; region_slots <= JREG_SLOTS = 128 KB of u32) */
    int64_t reg = blockIdx.x;
    if (reg >= nreg) return;
    __shared__ int32_t cnt[JREG_SLOTS];
    __shared__ int32_t wpart[TG_BLOCK / 64];
    int64_t sbase = reg * region_slots;
    const int per = (int)((region_slots + blockDim.x - 1) / blockDim.x);
    int32_t mysum = 0;
    for (int k = 0; k < per; k++) {
        int64_t sidx = (int64_t)threadIdx.x * per + k;
        if (sidx < region_slots) {
            int32_t v = bucket_cnt[sbase + sidx];
            cnt[sidx] = v;
            mysum += v;
        }
    }
    int32_t wpre = mysum;
    #pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
        int32_t o = __shfl_up(wpre, off, 64);
        if ((int)(threadIdx.x % 64) >= off) wpre += o;
    }
    if (threadIdx.x % 64 == 63) wpart[threadIdx.x / 64] = wpre;
    __syncthreads();
    if (threadIdx.x == 0) {
        int32_t run = 0;
        for (int w = 0; w < (int)(blockDim.x / 64); w++) {
            int32_t v = wpart[w];
            wpart[w] = run;
            run += v;
        }
        region_total[reg] = run;
    }
    __syncthreads();
    int32_t base = wpart[threadIdx.x / 64] + wpre - mysum;
    for (int k = 0; k < per; k++) {
        int64_t sidx = (int64_t)threadIdx.x * per + k;
        if (sidx >= region_slots) break;
        int32_t v = cnt[sidx];
        bucket_off[sbase + sidx] = base;
        base += v;
    }
}

__global__ void k_jc_addbase(int32_t* __restrict__ bucket_off, int64_t capacity,
                             const int32_t* __restrict__ region_base,
                             int64_t region_slots)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < capacity; i += stride)
        bucket_off[i] += region_base[i / region_slots];
}

__global__ void k_jc_scatter(JoinTable t, const uint32_t* __restrict__ slot_of,
                             int32_t* __restrict__ cursor /* copy of bucket_off */)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < t.n; i += stride) {
        uint32_t slot = slot_of[i];
        if (slot == 0xFFFFFFFFu) continue;      /* null key: never indexed */
        int32_t at = atomicAdd(&cursor[slot], 1);
        t.csr_rows[at] = (int32_t)i;
    }
}

__global__ void k_jc_sort(JoinTable t)
{
    /* ascending insertion sort within each bucket (expected length n/cap
     * < 1; the tail is a handful of duplicates sharing a slot) */
    int64_t s0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t sl = s0; sl < t.capacity; sl += stride) {
        int32_t lo = t.bucket_off[sl], hi = t.bucket_off[sl + 1];
        for (int32_t a = lo + 1; a < hi; a++) {
            int32_t v = t.csr_rows[a];
            int32_t b = a - 1;
            for (; b >= lo && t.csr_rows[b] > v; b--) t.csr_rows[b + 1] = t.csr_rows[b];
            t.csr_rows[b + 1] = v;
        }
    }
}

__global__ void k_csr_keys(JoinTable t, int64_t* __restrict__ ck)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t tot = t.bucket_off[t.capacity];   /* indexed rows (null keys excluded) */
    for (; i < tot; i += stride) ck[i] = t.keys[t.csr_rows[i]];
}

__global__ void k_build_slotkv(JoinTable t, SlotKV* __restrict__ kv)
{
    int64_t s0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t sl = s0; sl < t.capacity; sl += stride) {
        int32_t lo = t.bucket_off[sl], hi = t.bucket_off[sl + 1];
        SlotKV e;
        if (hi - lo == 1) { e.row = t.csr_rows[lo]; e.key = t.keys[e.row]; }
        else { e.row = (hi == lo) ? -1 : -2; e.key = 0; }
        e._pad = 0;
        kv[sl] = e;
    }
}

__global__ void k_set_bits(const int64_t* __restrict__ keys,
                           const uint64_t* __restrict__ valid, int64_t n,
                           int64_t base, uint64_t* __restrict__ bm)
{
    /* wave-segmented: clustered keys (lineitem orderkeys) put many lanes in
     * the same bitmap WORD — OR-combine per equal-word run in registers and
     * let only the run's last lane issue the atomic (Q4's 380M-key build:
     * per-lane check-then-set atomics measured 18.3 ms, ~10x the traffic) */
    int lane = threadIdx.x % 64;
    int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t iw = i0 - lane; iw < n; iw += stride) {
        int64_t i = iw + lane;
        bool active = i < n &&
                      !(valid && !((valid[i >> 6] >> (i & 63)) & 1));
        int64_t w = -1;
        uint64_t bit = 0;
        if (active) {
            uint64_t k = (uint64_t)(keys[i] - base);
            w = (int64_t)(k >> 6);
            bit = 1ull << (k & 63);
        }
        /* inclusive OR-scan over equal-w prefixes */
        #pragma unroll
        for (int off = 1; off < 64; off <<= 1) {
            uint64_t ob = __shfl_up(bit, off, 64);
            int64_t ow = __shfl_up(w, off, 64);
            if (lane >= off && ow == w) bit |= ob;
        }
        int64_t wnext = __shfl_down(w, 1, 64);
        bool last = active && (lane == 63 || i + 1 >= n || wnext != w);
        if (last && (bm[w] & bit) != bit)
            atomicOr((unsigned long long*)&bm[w], (unsigned long long)bit);
    }
}

__global__ void k_join_build(JoinTable t)
{
    /* Probe with PLAIN loads and resolve races with device-scope atomics:
     * a stale -1 is corrected by the CAS (which returns the real occupant),
     * and a slot's occupant KEY identity never changes after the first claim
     * (only equal-key rows are exchanged in), so a stale occupant row index
     * still compares the right key. Agent-scope atomic LOADS per probe step
     * bypass the caches and measured 15 ms per 15M-row build; plain loads
     * with CAS fallback are ~an order of magnitude cheaper. */
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < t.n; i += stride) {
        if (t.key_valid && !((t.key_valid[i >> 6] >> (i & 63)) & 1)) continue;
        int64_t key = t.keys[i];
        int64_t slot = (int64_t)(d_murmur3_mix((uint64_t)key) & (uint64_t)t.mask);
        while (true) {
            int32_t cur = *(volatile int32_t*)&t.slots[slot];
            if (cur == -1) {
                int32_t expected = -1;
                if (__hip_atomic_compare_exchange_strong(&t.slots[slot], &expected,
                        (int32_t)i, __ATOMIC_ACQ_REL, __ATOMIC_RELAXED,
                        __HIP_MEMORY_SCOPE_AGENT))
                    break;
                cur = expected;
            }
            if (t.keys[cur] == key) {
                /* chain push: links[mine] = previous head (ArrayPositionLinks) */
                int32_t prev = __hip_atomic_exchange(&t.slots[slot], (int32_t)i,
                        __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
                t.links[i] = prev;
                break;
            }
            slot = (slot + 1) & t.mask;
        }
    }
}

/* probe phase 1 (DefaultPagesHash.getAddressIndex hash-all/gather-all/verify
 * shape): count matches per probe row, caching the matched head so the fill
 * pass never re-probes the table */
/* ---- partitioned single-pass probe (large single-BIGINT-key tables) ----
 * The two-pass count/fill walk reads 2-3 random cache lines per probe over a
 * table far larger than L2. Instead: bucket-partition the PROBE ROWS by slot
 * region (so consecutive rows hit a table slice that stays cache-resident),
 * walk each bucket once appending (probe_row, build_row) matches, then
 * restore probe-row order with a stable radix sort — the reference emits
 * matches in probe position order (DefaultPageJoiner), duplicates in
 * reverse-insertion order (descending walk + stable sort preserves it). */
__global__ void k_pb_slots(JoinTable t, ProbeKeys p, int64_t m, int64_t nparts,
                           int shift, uint32_t* __restrict__ slot_of,
                           int32_t* __restrict__ pcount)
{
    extern __shared__ int32_t lh[];
    for (int64_t k = threadIdx.x; k < nparts; k += blockDim.x) lh[k] = 0;
    __syncthreads();
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < m; i += stride) {
        uint32_t slot = probe_null(t, p, i) ? 0xFFFFFFFFu : probe_slot(t, p, i);
        slot_of[i] = slot;
        if (slot != 0xFFFFFFFFu) atomicAdd(&lh[slot >> shift], 1);
    }
    __syncthreads();
    for (int64_t k = threadIdx.x; k < nparts; k += blockDim.x)
        if (lh[k]) atomicAdd(&pcount[k], lh[k]);
}

__global__ void k_pb_scatter(JoinTable t, ProbeKeys p, int64_t m, int shift,
                             const uint32_t* __restrict__ slot_of,
                             int32_t* __restrict__ pcur, int64_t nparts,
                             uint32_t* __restrict__ pi, uint32_t* __restrict__ pslot,
                             int64_t* __restrict__ pkey)
{
    /* block-hierarchical: LDS per-partition counts -> one global reserve per
     * (block, partition) -> LDS-cursor scatter. A flat per-row atomicAdd on
     * the handful of partition cursors serialized the whole pass (measured
     * 1.3 s/step at 275M rows x 32 partitions). */
    extern __shared__ int32_t sh[];      /* [nparts] counts, then cursors */
    int64_t chunk = (m + gridDim.x - 1) / gridDim.x;
    int64_t lo = (int64_t)blockIdx.x * chunk, hi = min(lo + chunk, m);
    for (int64_t k = threadIdx.x; k < nparts; k += blockDim.x) sh[k] = 0;
    __syncthreads();
    for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t slot = slot_of[i];
        if (slot != 0xFFFFFFFFu) atomicAdd(&sh[slot >> shift], 1);
    }
    __syncthreads();
    for (int64_t k = threadIdx.x; k < nparts; k += blockDim.x) {
        int32_t c = sh[k];
        sh[k] = c ? atomicAdd(&pcur[k], c) : 0;
    }
    __syncthreads();
    for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t slot = slot_of[i];
        if (slot == 0xFFFFFFFFu) continue;
        int32_t at = atomicAdd(&sh[slot >> shift], 1);
        pi[at] = (uint32_t)i;
        pslot[at] = slot;
        pkey[at] = p.pk[i];
    }
}

__global__ void k_pb_probe(JoinTable t, int64_t jlo, int64_t jhi,
                           const uint32_t* __restrict__ pi,
                           const uint32_t* __restrict__ pslot,
                           const int64_t* __restrict__ pkey,
                           unsigned long long* __restrict__ out_cnt,
                           int64_t cap,
                           uint32_t* __restrict__ mp, int32_t* __restrict__ mb)
{
    /* ONE LAUNCH PER PARTITION (host loop, stream-ordered): while this
     * launch runs, every bucket_off/csr_keys/csr_rows read lands in the
     * partition's ~16 MB table slice, which stays cache-resident.
     * DESCENDING walk + per-thread ordered appends + the later stable sort
     * by probe row = the reference's reverse-insertion duplicate order. */
    int64_t midx = jhi - jlo;
    int64_t chunk = (midx + gridDim.x - 1) / gridDim.x;
    int64_t lo = jlo + (int64_t)blockIdx.x * chunk, hi = min(lo + chunk, jhi);
    int lane = threadIdx.x % 64;
    /* wave-aggregated reservation: one atomicAdd per wave iteration instead
     * of one per match (a single global counter otherwise serializes) */
    for (int64_t j0 = lo + (threadIdx.x / 64) * 64; j0 < hi;
         j0 += (int64_t)(blockDim.x / 64) * 64 * 1) {
        int64_t j = j0 + lane;
        int32_t cnt = 0;
        uint32_t slot = 0; int64_t key = 0;
        int32_t blo = 0, bhi = 0;
        if (j < hi) {
            slot = pslot[j];
            key = pkey[j];
            blo = t.bucket_off[slot];
            bhi = t.bucket_off[slot + 1];
            for (int32_t x = bhi - 1; x >= blo; x--) cnt += (t.csr_keys[x] == key);
        }
        int32_t pre = cnt;
        #pragma unroll
        for (int off = 1; off < 64; off <<= 1) {
            int32_t o = __shfl_up(pre, off, 64);
            if (lane >= off) pre += o;
        }
        int32_t wave_total = __shfl(pre, 63, 64);
        unsigned long long base = 0;
        if (lane == 63 && wave_total)
            base = atomicAdd(out_cnt, (unsigned long long)wave_total);
        base = __shfl(base, 63, 64);
        if (j < hi && cnt) {
            int64_t pos = (int64_t)base + pre - cnt;
            for (int32_t x = bhi - 1; x >= blo; x--) {
                if (t.csr_keys[x] == key && pos < cap) {
                    mp[pos] = pi[j];
                    mb[pos] = t.csr_rows[x];
                    pos++;
                }
            }
        }
    }
}

__global__ void k_probe_count(JoinTable t, ProbeKeys p, int64_t m, int outer,
                              int32_t* __restrict__ counts,
                              int32_t* __restrict__ heads)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < m; i += stride) {
        int32_t cnt = 0;
        int32_t head = -1;
        if (!probe_null(t, p, i)) {
            int64_t slot = (int64_t)probe_slot(t, p, i);
            if (t.kv) {          /* one 16B read resolves most probes */
                SlotKV e = t.kv[slot];
                if (e.row >= 0) {
                    head = (e.key == p.pk[i]) ? -2 - e.row : -1;
                    cnt = (head != -1);
                }
                else if (e.row == -1) head = -1;
                else {               /* multi-row bucket: CSR walk */
                    int32_t matched = -1;
                    for (int32_t x = t.bucket_off[slot]; x < t.bucket_off[slot + 1]; x++) {
                        int32_t row = t.csr_rows[x];
                        if (probe_matches(t, p, i, row)) { cnt++; matched = row; }
                    }
                    head = (cnt == 1) ? -2 - matched : (cnt == 0 ? -1 : (int32_t)slot);
                }
            }
            else if (t.csr && t.csr_keys) {   /* single-key: slot-local compare */
                int64_t key = p.pk[i];
                int32_t matched = -1;
                for (int32_t x = t.bucket_off[slot]; x < t.bucket_off[slot + 1]; x++) {
                    if (t.csr_keys[x] == key) { cnt++; matched = x; }
                }
                head = (cnt == 1) ? -2 - t.csr_rows[matched]
                                  : (cnt == 0 ? -1 : (int32_t)slot);
            }
            else if (t.csr) {
                int32_t matched = -1;
                for (int32_t x = t.bucket_off[slot]; x < t.bucket_off[slot + 1]; x++) {
                    int32_t row = t.csr_rows[x];
                    if (probe_matches(t, p, i, row)) { cnt++; matched = row; }
                }
                /* single match (the usual case: unique build keys): encode
                 * the row as -2-row so the fill pass skips the bucket walk */
                head = (cnt == 1) ? -2 - matched : (cnt == 0 ? -1 : (int32_t)slot);
            }
            else {   /* legacy open addressing: single-BIGINT only */
                int64_t key = p.pk[i];
                int32_t cur;
                while ((cur = t.slots[slot]) != -1) {
                    if (t.keys[cur] == key) {
                        head = cur;
                        for (int32_t q = cur; q != -1; q = t.links[q]) cnt++;
                        break;
                    }
                    slot = (slot + 1) & t.mask;
                }
            }
        }
        /* probe-outer (LEFT): unmatched rows (incl. null keys) emit one
         * row with a null build side (LookupJoinOperator probeOnOuterSide,
         * LookupJoinOperators.java) */
        if (outer && cnt == 0) { cnt = 1; head = INT32_MIN; }
        counts[i] = cnt;
        heads[i] = head;
    }
}

/* probe phase 2: emit (probe_row, build_row) pairs from the cached heads.
 * CSR: walk the bucket DESCENDING — scatter order is global row order, so
 * this reproduces ArrayPositionLinks' reverse-insertion emission exactly
 * (and deterministically). */
__global__ void k_probe_fill(JoinTable t, ProbeKeys p, int64_t m,
                             const int32_t* __restrict__ heads,
                             const int64_t* __restrict__ offsets,
                             int32_t* __restrict__ out_probe,
                             int32_t* __restrict__ out_build)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < m; i += stride) {
        int64_t at = offsets[i];
        int32_t h = heads[i];
        if (h == -1) continue;
        if (h == INT32_MIN) {    /* probe-outer miss: null build side */
            out_probe[at] = (int32_t)i;
            out_build[at] = -1;
            continue;
        }
        if (h <= -2) {           /* pre-resolved single match */
            out_probe[at] = (int32_t)i;
            out_build[at] = -2 - h;
            continue;
        }
        if (t.csr && t.csr_keys) {
            int64_t key = p.pk[i];
            for (int32_t x = t.bucket_off[h + 1] - 1; x >= t.bucket_off[h]; x--) {
                if (t.csr_keys[x] == key) {
                    out_probe[at] = (int32_t)i;
                    out_build[at] = t.csr_rows[x];
                    at++;
                }
            }
        }
        else if (t.csr) {
            for (int32_t x = t.bucket_off[h + 1] - 1; x >= t.bucket_off[h]; x--) {
                int32_t row = t.csr_rows[x];
                if (probe_matches(t, p, i, row)) {
                    out_probe[at] = (int32_t)i;
                    out_build[at] = row;
                    at++;
                }
            }
        }
        else {
            for (int32_t q = h; q != -1; q = t.links[q]) {
                out_probe[at] = (int32_t)i;
                out_build[at] = q;
                at++;
            }
        }
    }
}

/* two-level exclusive scan of per-row counts (n can be 10^8+):
 * chunk sums -> serial chunk scan -> per-row offsets within chunk */
#define JSCAN_CHUNK 65536
__global__ void k_scan_chunk_sums(const int32_t* __restrict__ counts, int64_t n,
                                  int64_t* __restrict__ chunk_sums, int64_t nchunks)
{
    /* one wave per chunk: lanes stride the chunk, shuffle-reduce */
    int64_t c = (int64_t)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    if (c >= nchunks) return;
    int lane = threadIdx.x % 64;
    int64_t lo = c * JSCAN_CHUNK, hi = min(lo + JSCAN_CHUNK, n);
    long long s = 0;
    for (int64_t i = lo + lane; i < hi; i += 64) s += counts[i];
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1) s += __shfl_xor(s, off, 64);
    if (lane == 0) chunk_sums[c] = s;
}

__global__ void k_scan_chunks_serial(int64_t* chunk_sums, int64_t nchunks, int64_t* total)
{
    if (blockIdx.x || threadIdx.x) return;
    int64_t run = 0;
    for (int64_t i = 0; i < nchunks; i++) {
        int64_t v = chunk_sums[i];
        chunk_sums[i] = run;
        run += v;
    }
    *total = run;
}

__global__ void k_scan_row_offsets(const int32_t* __restrict__ counts, int64_t n,
                                   const int64_t* __restrict__ chunk_sums,
                                   int64_t* __restrict__ offsets, int64_t nchunks)
{
    /* one wave per chunk walks it carrying a running sum */
    int64_t c = (int64_t)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    if (c >= nchunks) return;
    int lane = threadIdx.x % 64;
    int64_t lo = c * JSCAN_CHUNK, hi = min(lo + JSCAN_CHUNK, n);
    int64_t run = chunk_sums[c];
    for (int64_t g = lo; g < hi; g += 64) {
        int64_t i = g + lane;
        long long v = (i < hi) ? counts[i] : 0;
        /* exclusive wave prefix via shuffles */
        long long pre = v;
        #pragma unroll
        for (int off = 1; off < 64; off <<= 1) {
            long long o = __shfl_up(pre, off, 64);
            if (lane >= off) pre += o;
        }
        long long wave_total = __shfl(pre, 63, 64);
        if (i < hi) offsets[i] = run + pre - v;
        run += wave_total;
    }
}

static tg_status run_scan_counts(tg_session* s, const int32_t* d_counts, int64_t n,
                                 int64_t* d_offsets, int64_t* d_total)
{
    int64_t nchunks = (n + JSCAN_CHUNK - 1) / JSCAN_CHUNK;
    if (nchunks < 1) nchunks = 1;
    int64_t* d_cs = nullptr;
    TG_POOL_ALLOC(s, &d_cs, nchunks * 8);
    int swpb = TG_BLOCK / 64;
    hipLaunchKernelGGL(k_scan_chunk_sums, dim3((uint32_t)((nchunks + swpb - 1) / swpb)),
                       dim3(TG_BLOCK), 0, s->stream, d_counts, n, d_cs, nchunks);
    TG_HIP_CHECK(hipGetLastError());
    hipLaunchKernelGGL(k_scan_chunks_serial, dim3(1), dim3(1), 0, s->stream,
                       d_cs, nchunks, d_total);
    TG_HIP_CHECK(hipGetLastError());
    int wpb = TG_BLOCK / 64;
    hipLaunchKernelGGL(k_scan_row_offsets, dim3((uint32_t)((nchunks + wpb - 1) / wpb)),
                       dim3(TG_BLOCK), 0, s->stream, d_counts, n, d_cs, d_offsets, nchunks);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_cs);
    return TG_OK;
}

__global__ void k_bitmap_concat(const uint64_t* __restrict__ src, int64_t n,
                                int64_t at, uint64_t* __restrict__ dst)
{
    /* dst preset to all-valid; clear dst bit (at+i) where src row i is null.
     * Works at any bit offset (the memcpy path needs 64-row alignment). */
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        if (!((src[i >> 6] >> (i & 63)) & 1)) {
            int64_t r = at + i;
            atomicAnd((unsigned long long*)&dst[r >> 6], ~(1ull << (r & 63)));
        }
    }
}

__global__ void k_off_rebase(const int32_t* __restrict__ src, int64_t n,
                             int32_t base, int32_t* __restrict__ dst)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) dst[i] = src[i] + base;
}

/* ---- build operator ---- */
struct HashBuilderOp : tg_operator {
    tg_join_bridge* bridge = nullptr;
    bool set_only = false;          /* SetBuilderOperator: bitmap, no index */
    bool range_done_ = false;

    tg_status compute_key_range()
    {
        JoinTable& t = bridge->t;
        long long* d_mm = nullptr;
        TG_POOL_ALLOC(s, &d_mm, 3 * 8);
        long long init[3] = {INT64_MAX, INT64_MIN, 0};
        TG_HIP_CHECK(hipMemcpyAsync(d_mm, init, 24, hipMemcpyHostToDevice, s->stream));
        if (total_rows > 0) {
            hipLaunchKernelGGL(k_key_minmax, dim3(tg_grid_for(total_rows)), dim3(TG_BLOCK),
                               0, s->stream, t.keys, t.key_valid, total_rows,
                               d_mm, d_mm + 1, (unsigned long long*)(d_mm + 2));
            TG_HIP_CHECK(hipGetLastError());
        }
        long long mm[3];
        TG_HIP_CHECK(hipMemcpyAsync(mm, d_mm, 24, hipMemcpyDeviceToHost, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_mm);
        bridge->key_min = mm[0]; bridge->key_max = mm[1]; bridge->key_rows = mm[2];
        range_done_ = true;
        return TG_OK;
    }
    std::vector<int32_t> key_channels;
    std::vector<tg_type> types;
    std::vector<DevPage> pages;     /* PagesIndex */
    int64_t total_rows = 0;

    /* concatenate one VARCHAR channel across the accumulated pages
     * (PagesIndex keeps per-page VariableWidthBlocks; the flat build-side
     * copy rebases each page's offsets onto a single byte array) */
    tg_status concat_varchar(size_t c, DevBlock* b)
    {
        int64_t total_bytes = 0;
        std::vector<int32_t> page_bytes(pages.size());
        for (size_t pi = 0; pi < pages.size(); pi++) {
            const DevBlock& pb = pages[pi].blocks[c];
            int32_t tb = 0;
            if (pb.n) {
                TG_HIP_CHECK(hipMemcpy(&tb, pb.offsets + pb.n, 4,
                                       hipMemcpyDeviceToHost));
            }
            page_bytes[pi] = tb;
            total_bytes += tb;
        }
        TG_POOL_ALLOC(s, &b->data, total_bytes ? total_bytes : 1);
        TG_POOL_ALLOC(s, &b->offsets, (total_rows + 1) * 4);
        bool anynull = false;
        for (auto& p : pages) anynull |= p.blocks[c].valid != nullptr;
        if (anynull) {
            int64_t words = (total_rows + 63) / 64;
            TG_POOL_ALLOC(s, &b->valid, words * 8);
            TG_HIP_CHECK(hipMemsetAsync(b->valid, 0xFF, words * 8, s->stream));
        }
        int64_t at = 0, byte_at = 0;
        for (size_t pi = 0; pi < pages.size(); pi++) {
            const DevBlock& pb = pages[pi].blocks[c];
            if (page_bytes[pi]) {
                TG_HIP_CHECK(hipMemcpyAsync((char*)b->data + byte_at, pb.data,
                                            page_bytes[pi],
                                            hipMemcpyDeviceToDevice, s->stream));
            }
            if (pb.n) {
                hipLaunchKernelGGL(k_off_rebase, dim3(tg_grid_for(pb.n)),
                                   dim3(TG_BLOCK), 0, s->stream,
                                   pb.offsets, pb.n, (int32_t)byte_at,
                                   b->offsets + at);
                TG_HIP_CHECK(hipGetLastError());
            }
            if (pb.valid) {
                hipLaunchKernelGGL(k_bitmap_concat, dim3(tg_grid_for(pb.n)),
                                   dim3(TG_BLOCK), 0, s->stream, pb.valid, pb.n,
                                   at, b->valid);
                TG_HIP_CHECK(hipGetLastError());
            }
            at += pb.n;
            byte_at += page_bytes[pi];
        }
        int32_t tb32 = (int32_t)total_bytes;
        TG_HIP_CHECK(hipMemcpyAsync(b->offsets + total_rows, &tb32, 4,
                                    hipMemcpyHostToDevice, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        return TG_OK;
    }

    tg_status add_input(const tg_page* page) override
    {
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        total_rows += in.n;
        pages.emplace_back(std::move(in));
        return TG_OK;
    }

    tg_status finish() override
    {
        if (input_finished) return TG_OK;
        input_finished = true;
        /* concatenate channels (contiguous by global build row) + key copy */
        JoinTable& t = bridge->t;
        t.n = total_rows;
        t.capacity = join_hash_size(total_rows);
        t.mask = t.capacity - 1;
        /* probe kernels encode a multi-match bucket slot as (int32_t)slot
         * (k_probe_count heads); a table beyond INT32_MAX slots would wrap
         * negative and be read back as a miss — reject at build, mirroring
         * BigintGroupByHash's own 2^30 capacity cap (BigintGroupByHash.java) */
        if (t.capacity > (int64_t)INT32_MAX) {
            TG_SET_ERR("join build of %lld rows needs %lld slots > INT32_MAX",
                       (long long)total_rows, (long long)t.capacity);
            return TG_ERR_OOM;
        }
        static int pre_csr = [] { const char* e = getenv("TG_JOIN_CSR"); return e ? atoi(e) : 1; }();
        if (!(pre_csr && total_rows > 0)) {
            TG_POOL_ALLOC(s, &t.slots, t.capacity * 4);
            TG_POOL_ALLOC(s, &t.links, (total_rows ? total_rows : 1) * 4);
        }
        TG_POOL_ALLOC(s, &t.keys, (total_rows ? total_rows : 1) * 8);
        bool any_key_null = false;
        for (auto& p : pages)
            if (!p.blocks.empty() && p.blocks[key_channels[0]].valid) any_key_null = true;
        if (any_key_null) {
            int64_t words = (total_rows + 63) / 64;
            TG_POOL_ALLOC(s, &t.key_valid, words * 8);
            TG_HIP_CHECK(hipMemsetAsync(t.key_valid, 0xFF, words * 8, s->stream));
        }
        /* concat all channels */
        bridge->build_channels.resize(types.size());
        for (size_t c = 0; c < types.size(); c++) {
            DevBlock& b = bridge->build_channels[c];
            b.type = types[c];
            b.n = total_rows;
            if (types[c] == TG_VARCHAR) {
                tg_status vst = concat_varchar(c, &b);
                if (vst != TG_OK) return vst;
                continue;
            }
            TG_POOL_ALLOC(s, &b.data, (total_rows ? total_rows : 1) * b.elem_size());
            int64_t at = 0;
            bool anynull = false;
            for (auto& p : pages) anynull |= p.blocks[c].valid != nullptr;
            if (anynull) {
                int64_t words = (total_rows + 63) / 64;
                TG_POOL_ALLOC(s, &b.valid, words * 8);
                TG_HIP_CHECK(hipMemsetAsync(b.valid, 0xFF, words * 8, s->stream));
            }
            for (auto& p : pages) {
                TG_HIP_CHECK(hipMemcpyAsync((char*)b.data + at * b.elem_size(),
                                            p.blocks[c].data, p.n * b.elem_size(),
                                            hipMemcpyDeviceToDevice, s->stream));
                /* null bitmaps at arbitrary bit offsets: pages are
                 * TG_BLOCK-row-aligned in practice; general repack via gather
                 * kernel would go here — round 1 requires page sizes to be
                 * multiples of 64 when nulls are present */
                if (p.blocks[c].valid) {
                    hipLaunchKernelGGL(k_bitmap_concat, dim3(tg_grid_for(p.n)),
                                       dim3(TG_BLOCK), 0, s->stream, p.blocks[c].valid,
                                       p.n, at, b.valid);
                    TG_HIP_CHECK(hipGetLastError());
                }
                at += p.n;
            }
        }
        /* single BIGINT key -> BigintPagesHash-style flat key copy;
         * otherwise the DefaultPagesHash-style generic channel compare
         * (requires the CSR table; the legacy open-addressing path stays
         * single-key). */
        t.generic = !(key_channels.size() == 1 &&
                      bridge->build_channels[key_channels[0]].type == TG_BIGINT);
        t.n_key_ch = (int32_t)key_channels.size();
        if (t.generic) {
            DevPage view;   /* non-owning view over the concatenated channels */
            view.n = total_rows;
            view.blocks = bridge->build_channels;
            for (auto& b : view.blocks) b.owned = false;
            tg_status kst = make_kcols(s, view, key_channels.data(),
                                       (int)key_channels.size(), &bridge->d_bkeys);
            view.blocks.clear();
            if (kst != TG_OK) return kst;
            t.bkeys = bridge->d_bkeys;
        }
        else {
            const DevBlock& kb = bridge->build_channels[key_channels[0]];
            TG_HIP_CHECK(hipMemcpyAsync(t.keys, kb.data, total_rows * 8,
                                        hipMemcpyDeviceToDevice, s->stream));
            if (t.key_valid && kb.valid) {
                TG_HIP_CHECK(hipMemcpyAsync(t.key_valid, kb.valid,
                                            (total_rows + 63) / 64 * 8,
                                            hipMemcpyDeviceToDevice, s->stream));
            }
        }
        /* A/B on MI355X (Q3 SF100): CSR 98.0 ms/step vs open addressing
         * 110.1 — CSR default; TG_JOIN_CSR=0 selects the legacy path */
        static int use_csr = [] { const char* e = getenv("TG_JOIN_CSR"); return e ? atoi(e) : 1; }();
        if (t.generic && !use_csr) { TG_SET_ERR("generic join keys require the CSR table"); return TG_ERR_UNSUPPORTED; }
        /* SetBuilderOperator path: dense-range membership bitmap replaces
         * the positional index entirely (the 380M-row Q4 build spent ~63 ms
         * in CSR construction a semi join never needs) */
        bool bitmap_built = false;
        if (set_only && !t.generic && total_rows > 0) {
            tg_status mst = compute_key_range();
            if (mst != TG_OK) return mst;
            int64_t range = bridge->key_rows > 0
                          ? bridge->key_max - bridge->key_min + 1 : 0;
            if (range > 0 && range <= (1ll << 33)) {   /* <= 1 GiB bitmap */
                int64_t words = (range + 63) / 64;
                uint64_t* bm = nullptr;
                TG_POOL_ALLOC(s, &bm, words * 8);
                TG_HIP_CHECK(hipMemsetAsync(bm, 0, words * 8, s->stream));
                hipLaunchKernelGGL(k_set_bits, dim3(tg_grid_for(total_rows)),
                                   dim3(TG_BLOCK), 0, s->stream, t.keys,
                                   t.key_valid, total_rows, bridge->key_min, bm);
                TG_HIP_CHECK(hipGetLastError());
                TG_HIP_CHECK(hipStreamSynchronize(s->stream));
                t.set_bitmap = bm;
                t.set_min = bridge->key_min;
                t.set_max = bridge->key_max;
                t.has_null_key = bridge->key_rows < total_rows ? 1 : 0;
                bitmap_built = true;
            }
        }
        /* dynamic-filter bitmap alongside the real index (requested via
         * tg_join_bridge_request_bitmap; single int64 key, dense range) */
        if (!set_only && bridge->want_bitmap && !t.generic && total_rows > 0) {
            tg_status mst = compute_key_range();
            if (mst != TG_OK) return mst;
            int64_t range = bridge->key_rows > 0
                          ? bridge->key_max - bridge->key_min + 1 : 0;
            if (range > 0 && range <= (1ll << 33)) {
                int64_t words = (range + 63) / 64;
                uint64_t* bm = nullptr;
                TG_POOL_ALLOC(s, &bm, words * 8);
                TG_HIP_CHECK(hipMemsetAsync(bm, 0, words * 8, s->stream));
                hipLaunchKernelGGL(k_set_bits, dim3(tg_grid_for(total_rows)),
                                   dim3(TG_BLOCK), 0, s->stream, t.keys,
                                   t.key_valid, total_rows, bridge->key_min, bm);
                TG_HIP_CHECK(hipGetLastError());
                TG_HIP_CHECK(hipStreamSynchronize(s->stream));
                t.set_bitmap = bm;
                t.set_min = bridge->key_min;
                t.set_max = bridge->key_max;
            }
        }
        if (!bitmap_built && use_csr && total_rows > 0) {
            t.csr = 1;
            int64_t region_slots = t.capacity < JREG_SLOTS ? t.capacity : JREG_SLOTS;
            int64_t nreg = t.capacity / region_slots;
            TG_POOL_ALLOC(s, &t.bucket_off, (t.capacity + 1) * 4);
            TG_POOL_ALLOC(s, &t.csr_rows, total_rows * 4);
            uint32_t* d_slot_of = nullptr;
            int32_t* d_cnt = nullptr;       /* per-slot counts, then cursors */
            int32_t* d_rtotal = nullptr;
            int32_t* d_rbase = nullptr;
            TG_POOL_ALLOC(s, &d_slot_of, total_rows * 4);
            TG_POOL_ALLOC(s, &d_cnt, t.capacity * 4);
            TG_POOL_ALLOC(s, &d_rtotal, nreg * 4);
            TG_POOL_ALLOC(s, &d_rbase, nreg * 4);
            TG_HIP_CHECK(hipMemsetAsync(d_cnt, 0, t.capacity * 4, s->stream));
            hipLaunchKernelGGL(k_jc_count, dim3(tg_grid_for(total_rows)), dim3(TG_BLOCK),
                               0, s->stream, t, d_slot_of, d_cnt);
            TG_HIP_CHECK(hipGetLastError());
            hipLaunchKernelGGL(k_jc_local_off, dim3((uint32_t)nreg), dim3(TG_BLOCK),
                               0, s->stream, d_cnt, t.bucket_off, d_rtotal,
                               nreg, region_slots);
            TG_HIP_CHECK(hipGetLastError());
            std::vector<int32_t> rt(nreg), rb(nreg);
            TG_HIP_CHECK(hipMemcpyAsync(rt.data(), d_rtotal, nreg * 4,
                                        hipMemcpyDeviceToHost, s->stream));
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            int32_t run = 0;
            for (int64_t r = 0; r < nreg; r++) { rb[r] = run; run += rt[r]; }
            int32_t total_indexed = run;    /* == total_rows minus null keys */
            t.has_null_key = (int64_t)total_indexed < total_rows ? 1 : 0;
            TG_HIP_CHECK(hipMemcpyAsync(d_rbase, rb.data(), nreg * 4,
                                        hipMemcpyHostToDevice, s->stream));
            hipLaunchKernelGGL(k_jc_addbase, dim3(tg_grid_for(t.capacity)), dim3(TG_BLOCK),
                               0, s->stream, t.bucket_off, t.capacity, d_rbase,
                               region_slots);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipMemcpyAsync(t.bucket_off + t.capacity, &total_indexed, 4,
                                        hipMemcpyHostToDevice, s->stream));
            TG_HIP_CHECK(hipMemcpyAsync(d_cnt, t.bucket_off, t.capacity * 4,
                                        hipMemcpyDeviceToDevice, s->stream));
            hipLaunchKernelGGL(k_jc_scatter, dim3(tg_grid_for(total_rows)), dim3(TG_BLOCK),
                               0, s->stream, t, d_slot_of, d_cnt);
            TG_HIP_CHECK(hipGetLastError());
            hipLaunchKernelGGL(k_jc_sort, dim3(tg_grid_for(t.capacity)), dim3(TG_BLOCK),
                               0, s->stream, t);
            TG_HIP_CHECK(hipGetLastError());
            if (!t.generic && total_rows > 0) {
                TG_POOL_ALLOC(s, &t.csr_keys, total_rows * 8);
                hipLaunchKernelGGL(k_csr_keys, dim3(tg_grid_for(total_rows)),
                                   dim3(TG_BLOCK), 0, s->stream, t, t.csr_keys);
                TG_HIP_CHECK(hipGetLastError());
            }
            /* A/B on MI355X (Q3 SF100): the 512 MB kv array RAISED probe
             * time (43.1 vs 39.7 ms per 10 launches) — the CSR arrays it
             * replaces are smaller and L2-cache better. Kept behind
             * TG_JOIN_KV=1 for small-table workloads. */
            static int use_kv = [] { const char* e = getenv("TG_JOIN_KV"); return e ? atoi(e) : 0; }();
            if (!t.generic && use_kv) {
                SlotKV* d_kv = nullptr;
                TG_POOL_ALLOC(s, &d_kv, t.capacity * (int64_t)sizeof(SlotKV));
                hipLaunchKernelGGL(k_build_slotkv, dim3(tg_grid_for(t.capacity)),
                                   dim3(TG_BLOCK), 0, s->stream, t, d_kv);
                TG_HIP_CHECK(hipGetLastError());
                t.kv = d_kv;
            }
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            tg_pool_free(s, d_slot_of);
            tg_pool_free(s, d_cnt);
            tg_pool_free(s, d_rtotal);
            tg_pool_free(s, d_rbase);
        }
        else if (!bitmap_built) {
            hipLaunchKernelGGL(k_join_init, dim3(tg_grid_for(t.capacity)), dim3(TG_BLOCK),
                               0, s->stream, t.slots, t.capacity, t.links, total_rows);
            TG_HIP_CHECK(hipGetLastError());
            if (total_rows > 0) {
                hipLaunchKernelGGL(k_join_build, dim3(tg_grid_for(total_rows)), dim3(TG_BLOCK),
                                   0, s->stream, t);
                TG_HIP_CHECK(hipGetLastError());
            }
        }
        /* dynamic filter source: min/max over non-null build keys
         * (single-BIGINT-key builds; generic keys report no range) */
        if (!t.generic && !range_done_) {
            tg_status mst = compute_key_range();
            if (mst != TG_OK) return mst;
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        for (auto& p : pages) tg_free_page(s, &p);
        pages.clear();
        bridge->built = true;
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        out->channel_count = 0;
        out->position_count = 0;
        out->blocks = nullptr;
        *finished = input_finished ? 1 : 0;
        return TG_OK;
    }

    ~HashBuilderOp() override
    {
        for (auto& p : pages) tg_free_page(s, &p);
    }
};

/* ---- probe operator ---- */
struct LookupJoinOp : tg_operator {
    tg_join_bridge* bridge = nullptr;
    std::vector<int32_t> key_channels;
    std::vector<tg_type> probe_types;
    std::vector<int32_t> probe_output;
    bool outer = false;             /* probe-outer (LEFT) */

    tg_status add_input(const tg_page* page) override
    {
        if (!bridge->built) { TG_SET_ERR("lookup source not built (finish the builder first)"); return TG_ERR_STATE; }
        if (bridge->t.set_bitmap && !bridge->t.csr) { TG_SET_ERR("set-builder bridge supports semi join only"); return TG_ERR_UNSUPPORTED; }
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        JoinTable& t = bridge->t;
        ProbeKeys pkc{};
        KColH* d_pkg = nullptr;
        if (t.generic) {
            st = make_kcols(s, in, key_channels.data(), (int)key_channels.size(), &d_pkg);
            if (st != TG_OK) { tg_free_page(s, &in); return st; }
            pkc.pkg = d_pkg;
        }
        else {
            const DevBlock& kb = in.blocks[key_channels[0]];
            if (kb.type != TG_BIGINT) { tg_free_page(s, &in); TG_SET_ERR("single join key must be BIGINT (use multi-channel keys otherwise)"); return TG_ERR_UNSUPPORTED; }
            pkc.pk = (const int64_t*)kb.data;
            pkc.pvalid = kb.valid;
        }
        int32_t* d_counts = nullptr;
        int32_t* d_heads = nullptr;
        int64_t* d_offsets = nullptr;
        int64_t* d_total = nullptr;
        int32_t* d_op = nullptr;
        int32_t* d_ob = nullptr;
        int64_t total = 0;
        int64_t tbl_bytes = t.capacity * 4 + t.n * 12;
        /* A/B on MI355X (Q3 SF100): partitioned probe 41 ms/step vs classic
         * 28.6 — the ~300 MB table already rides the 256 MB LLC, so the
         * extra partition passes (5.6 GB) and match sort never pay back.
         * Kept behind TG_JOIN_PART=1 for tables that outgrow the LLC. */
        const char* ep = getenv("TG_JOIN_PART");
        int use_part = ep ? atoi(ep) : 0;
        const char* emr = getenv("TG_JOIN_PART_MIN_ROWS");
        int64_t min_rows = emr ? atoll(emr) : (1 << 22);
        const char* emb = getenv("TG_JOIN_PART_MIN_BYTES");
        int64_t min_bytes = emb ? atoll(emb) : (64ll << 20);
        if (use_part && !outer && t.csr && !t.generic && t.csr_keys &&
            in.n >= min_rows && tbl_bytes > min_bytes) {
            /* partitioned single-pass probe + stable match sort */
            int64_t nparts = 1;
            while (tbl_bytes / nparts > (16ll << 20)) nparts <<= 1;
            if (nparts > t.capacity) nparts = t.capacity;
            int shift = 0;
            while (((int64_t)1 << shift) < t.capacity / nparts) shift++;
            uint32_t* d_slot_of = nullptr;
            int32_t* d_pc = nullptr;
            uint32_t* d_pi = nullptr;
            uint32_t* d_ps = nullptr;
            int64_t* d_pk = nullptr;
            unsigned long long* d_cnt = nullptr;
            TG_POOL_ALLOC(s, &d_slot_of, in.n * 4);
            TG_POOL_ALLOC(s, &d_pc, (nparts + 1) * 4);
            TG_POOL_ALLOC(s, &d_pi, in.n * 4);
            TG_POOL_ALLOC(s, &d_ps, in.n * 4);
            TG_POOL_ALLOC(s, &d_pk, in.n * 8);
            TG_POOL_ALLOC(s, &d_cnt, 8);
            TG_HIP_CHECK(hipMemsetAsync(d_pc, 0, (nparts + 1) * 4, s->stream));
            TG_HIP_CHECK(hipMemsetAsync(d_cnt, 0, 8, s->stream));
            hipLaunchKernelGGL(k_pb_slots, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                               (size_t)nparts * 4, s->stream, t, pkc, in.n, nparts,
                               shift, d_slot_of, d_pc);
            TG_HIP_CHECK(hipGetLastError());
            std::vector<int32_t> pc(nparts), pb(nparts);
            TG_HIP_CHECK(hipMemcpyAsync(pc.data(), d_pc, nparts * 4,
                                        hipMemcpyDeviceToHost, s->stream));
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            int32_t run = 0;
            for (int64_t q = 0; q < nparts; q++) { pb[q] = run; run += pc[q]; }
            int64_t midx = run;               /* non-null probe rows */
            TG_HIP_CHECK(hipMemcpyAsync(d_pc, pb.data(), nparts * 4,
                                        hipMemcpyHostToDevice, s->stream));
            hipLaunchKernelGGL(k_pb_scatter, dim3(2048), dim3(TG_BLOCK),
                               (size_t)nparts * 4, s->stream, t, pkc, in.n, shift,
                               d_slot_of, d_pc, nparts, d_pi, d_ps, d_pk);
            TG_HIP_CHECK(hipGetLastError());
            /* first-try match cap (exact when build keys are unique, the
               usual case); k_pb_probe counts past it without writing, so an
               overflow is detected and retried with the exact size */
            int64_t cap = midx + t.n + 64;
            TG_POOL_ALLOC(s, &d_op, cap * 4);
            TG_POOL_ALLOC(s, &d_ob, cap * 4);
            for (int64_t q = 0; q < nparts; q++) {
                if (!pc[q]) continue;
                hipLaunchKernelGGL(k_pb_probe, dim3(1024), dim3(TG_BLOCK), 0, s->stream,
                                   t, (int64_t)pb[q], (int64_t)pb[q] + pc[q],
                                   d_pi, d_ps, d_pk, d_cnt, cap,
                                   (uint32_t*)d_op, d_ob);
                TG_HIP_CHECK(hipGetLastError());
            }
            unsigned long long cnt = 0;
            TG_HIP_CHECK(hipMemcpyAsync(&cnt, d_cnt, 8, hipMemcpyDeviceToHost, s->stream));
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            total = (int64_t)cnt;
            if (total > cap) {       /* many-to-many overflow: exact retry */
                tg_pool_free(s, d_op);
                tg_pool_free(s, d_ob);
                cap = total;
                TG_POOL_ALLOC(s, &d_op, cap * 4);
                TG_POOL_ALLOC(s, &d_ob, cap * 4);
                TG_HIP_CHECK(hipMemsetAsync(d_cnt, 0, 8, s->stream));
                for (int64_t q = 0; q < nparts; q++) {
                    if (!pc[q]) continue;
                    hipLaunchKernelGGL(k_pb_probe, dim3(1024), dim3(TG_BLOCK), 0, s->stream,
                                       t, (int64_t)pb[q], (int64_t)pb[q] + pc[q],
                                       d_pi, d_ps, d_pk, d_cnt, cap,
                                       (uint32_t*)d_op, d_ob);
                    TG_HIP_CHECK(hipGetLastError());
                }
                TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            }
            tg_status pst = run_sort_pairs_u32(s, (uint32_t*)d_op, d_ob, total);
            if (pst != TG_OK) return pst;
            tg_pool_free(s, d_slot_of);
            tg_pool_free(s, d_pc);
            tg_pool_free(s, d_pi);
            tg_pool_free(s, d_ps);
            tg_pool_free(s, d_pk);
            tg_pool_free(s, d_cnt);
        }
        else {
            TG_POOL_ALLOC(s, &d_counts, (in.n ? in.n : 1) * 4);
            TG_POOL_ALLOC(s, &d_heads, (in.n ? in.n : 1) * 4);
            TG_POOL_ALLOC(s, &d_offsets, (in.n ? in.n : 1) * 8);
            TG_POOL_ALLOC(s, &d_total, 8);
            hipLaunchKernelGGL(k_probe_count, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                               0, s->stream, t, pkc, in.n, outer ? 1 : 0,
                               d_counts, d_heads);
            TG_HIP_CHECK(hipGetLastError());
            tg_status sst = run_scan_counts(s, d_counts, in.n, d_offsets, d_total);
            if (sst != TG_OK) return sst;
            TG_HIP_CHECK(hipMemcpyAsync(&total, d_total, 8, hipMemcpyDeviceToHost, s->stream));
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            TG_POOL_ALLOC(s, &d_op, (total ? total : 1) * 4);
            TG_POOL_ALLOC(s, &d_ob, (total ? total : 1) * 4);
            hipLaunchKernelGGL(k_probe_fill, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                               0, s->stream, t, pkc, in.n, d_heads, d_offsets, d_op, d_ob);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        }

        /* output: probe output channels gathered by probe row, then build
         * output channels gathered by build row (LookupJoinPageBuilder) */
        DevPage outp;
        outp.n = total;
        for (int32_t ch : probe_output) {
            DevBlock ob;
            st = run_gather(s, in.blocks[ch], d_op, (int32_t)total, &ob);
            if (st != TG_OK) return st;
            outp.blocks.push_back(ob);
        }
        for (int32_t ch : bridge->build_output_channels) {
            DevBlock ob;
            st = run_gather(s, bridge->build_channels[ch], d_ob, (int32_t)total, &ob,
                            outer /* -1 positions become NULL */);
            if (st != TG_OK) return st;
            outp.blocks.push_back(ob);
        }
        if (d_counts) tg_pool_free(s, d_counts);
        if (d_heads) tg_pool_free(s, d_heads);
        if (d_offsets) tg_pool_free(s, d_offsets);
        if (d_total) tg_pool_free(s, d_total);
        if (d_pkg) tg_pool_free(s, d_pkg);
        tg_pool_free(s, d_op);
        tg_pool_free(s, d_ob);
        tg_free_page(s, &in);
        stage_output(std::move(outp));
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        emit_staged(out, finished);
        return TG_OK;
    }

    ~LookupJoinOp() override
    {
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

extern "C" tg_status tg_join_bridge_create(tg_session* s, tg_join_bridge** out)
{
    if (!s || !out) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    auto* b = new tg_join_bridge();
    b->s = s;
    *out = b;
    return TG_OK;
}

extern "C" void tg_join_bridge_close(tg_join_bridge* b)
{
    if (!b) return;
    if (b->t.slots) tg_pool_free(b->s, b->t.slots);
    if (b->t.links) tg_pool_free(b->s, b->t.links);
    if (b->t.keys) tg_pool_free(b->s, b->t.keys);
    if (b->t.key_valid) tg_pool_free(b->s, b->t.key_valid);
    if (b->t.bucket_off) tg_pool_free(b->s, b->t.bucket_off);
    if (b->t.csr_rows) tg_pool_free(b->s, b->t.csr_rows);
    if (b->t.csr_keys) tg_pool_free(b->s, b->t.csr_keys);
    if (b->t.kv) tg_pool_free(b->s, (void*)b->t.kv);
    if (b->t.set_bitmap) tg_pool_free(b->s, (void*)b->t.set_bitmap);
    if (b->d_bkeys) tg_pool_free(b->s, b->d_bkeys);
    for (auto& c : b->build_channels) {
        if (c.data) tg_pool_free(b->s, c.data);
        if (c.valid) tg_pool_free(b->s, c.valid);
    }
    delete b;
}

extern "C" tg_status tg_hash_builder_create(tg_session* s, tg_join_bridge* bridge,
    const int32_t* build_types, int32_t n_build_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* output_channels, int32_t n_output_channels,
    tg_operator** out);

/* SetBuilderOperator analog (semi-join source: ChannelSet membership only;
 * dense key ranges use a bitmap, sparse ranges fall back to the CSR index) */
extern "C" tg_status tg_set_builder_create(tg_session* s, tg_join_bridge* bridge,
    const int32_t* build_types, int32_t n_build_channels,
    int32_t key_channel, tg_operator** out)
{
    int32_t kc = key_channel;
    tg_status st = tg_hash_builder_create(s, bridge, build_types, n_build_channels,
                                          &kc, 1, nullptr, 0, out);
    if (st != TG_OK) return st;
    static_cast<HashBuilderOp*>(*out)->set_only = true;
    return TG_OK;
}

extern "C" tg_status tg_hash_builder_create(tg_session* s, tg_join_bridge* bridge,
    const int32_t* build_types, int32_t n_build_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* output_channels, int32_t n_output_channels,
    tg_operator** out)
{
    if (!s || !bridge || !build_types || !key_channels || n_key_channels < 1 ||
        n_key_channels > 7) {
        TG_SET_ERR("1..7 join key channels");
        return TG_ERR_INVALID_ARG;
    }
    auto* op = new HashBuilderOp();
    op->s = s;
    op->bridge = bridge;
    op->key_channels.assign(key_channels, key_channels + n_key_channels);
    for (int i = 0; i < n_build_channels; i++)
        op->types.push_back((tg_type)build_types[i]);
    bridge->build_types = op->types;
    bridge->build_output_channels.assign(output_channels, output_channels + n_output_channels);
    *out = op;
    return TG_OK;
}

/* ---- semi join (operator/HashSemiJoinOperator.java: appends a BOOLEAN
 * "matched" channel to the probe page; null probe keys yield NULL) ---- */
__global__ void k_semi_probe(JoinTable t, ProbeKeys p, int64_t m,
                             int8_t* __restrict__ match, uint64_t* __restrict__ mvalid)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < m; i += stride) {
        /* three-valued IN semantics (HashSemiJoinOperator.java:180-201):
         * probe null -> false if the build set is empty, else NULL;
         * miss against a set containing NULL -> NULL. */
        if (probe_null(t, p, i)) {
            match[i] = 0;
            if (t.n > 0)
                atomicAnd((unsigned long long*)&mvalid[i >> 6], ~(1ull << (i & 63)));
            continue;
        }
        int8_t hit = 0;
        if (t.set_bitmap) {
            int64_t pk = p.pk[i];
            if (pk >= t.set_min && pk <= t.set_max) {
                uint64_t k = (uint64_t)(pk - t.set_min);
                hit = (t.set_bitmap[k >> 6] >> (k & 63)) & 1 ? 1 : 0;
            }
            match[i] = hit;
            if (!hit && t.has_null_key)
                atomicAnd((unsigned long long*)&mvalid[i >> 6], ~(1ull << (i & 63)));
            continue;
        }
        int64_t slot = (int64_t)probe_slot(t, p, i);
        if (t.kv) {
            SlotKV e = t.kv[slot];
            if (e.row >= 0) hit = (e.key == p.pk[i]);
            else if (e.row == -2) {
                for (int32_t x = t.bucket_off[slot]; x < t.bucket_off[slot + 1] && !hit; x++)
                    hit = probe_matches(t, p, i, t.csr_rows[x]);
            }
        }
        else if (t.csr && t.csr_keys) {
            int64_t key = p.pk[i];
            for (int32_t x = t.bucket_off[slot]; x < t.bucket_off[slot + 1] && !hit; x++)
                hit = (t.csr_keys[x] == key);
        }
        else if (t.csr) {
            for (int32_t x = t.bucket_off[slot]; x < t.bucket_off[slot + 1] && !hit; x++)
                hit = probe_matches(t, p, i, t.csr_rows[x]);
        }
        else {
            int64_t key = p.pk[i];
            int32_t cur;
            while ((cur = t.slots[slot]) != -1) {
                if (t.keys[cur] == key) { hit = 1; break; }
                slot = (slot + 1) & t.mask;
            }
        }
        match[i] = hit;
        if (!hit && t.has_null_key)
            atomicAnd((unsigned long long*)&mvalid[i >> 6], ~(1ull << (i & 63)));
    }
}

struct SemiJoinOp : tg_operator {
    tg_join_bridge* bridge = nullptr;
    int32_t key_channel = 0;

    tg_status add_input(const tg_page* page) override
    {
        if (!bridge->built) { TG_SET_ERR("lookup source not built"); return TG_ERR_STATE; }
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        JoinTable& jt = bridge->t;
        ProbeKeys pkc{};
        KColH* d_pkg = nullptr;
        if (jt.generic) {
            int32_t kcs[1] = {key_channel};
            st = make_kcols(s, in, kcs, 1, &d_pkg);
            if (st != TG_OK) { tg_free_page(s, &in); return st; }
            pkc.pkg = d_pkg;
        }
        else {
            const DevBlock& kb0 = in.blocks[key_channel];
            if (kb0.type != TG_BIGINT) { tg_free_page(s, &in); TG_SET_ERR("semi join single key must be BIGINT"); return TG_ERR_UNSUPPORTED; }
            pkc.pk = (const int64_t*)kb0.data;
            pkc.pvalid = kb0.valid;
        }
        DevBlock mb;
        mb.type = TG_BOOLEAN;
        mb.n = in.n;
        TG_POOL_ALLOC(s, &mb.data, in.n ? in.n : 1);
        int64_t words = (in.n + 63) / 64;
        TG_POOL_ALLOC(s, &mb.valid, (words ? words : 1) * 8);
        TG_HIP_CHECK(hipMemsetAsync(mb.valid, 0xFF, words * 8, s->stream));
        hipLaunchKernelGGL(k_semi_probe, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                           0, s->stream, bridge->t, pkc, in.n, (int8_t*)mb.data, mb.valid);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        if (d_pkg) tg_pool_free(s, d_pkg);
        DevPage outp;
        outp.n = in.n;
        for (auto& b : in.blocks) { outp.blocks.push_back(b); }
        in.blocks.clear();                /* ownership moved */
        outp.blocks.push_back(mb);
        stage_output(std::move(outp));
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        emit_staged(out, finished);
        return TG_OK;
    }

    ~SemiJoinOp() override
    {
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

extern "C" tg_status tg_semi_join_create(tg_session* s, tg_join_bridge* bridge,
                                         int32_t key_channel, tg_operator** out)
{
    if (!s || !bridge || !out) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    auto* op = new SemiJoinOp();
    op->s = s;
    op->bridge = bridge;
    op->key_channel = key_channel;
    *out = op;
    return TG_OK;
}

extern "C" tg_status tg_lookup_join_create_ex(tg_session* s, tg_join_bridge* bridge,
    const int32_t* probe_types, int32_t n_probe_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* probe_output_channels, int32_t n_probe_output,
    int32_t join_type /* 0 inner, 1 probe-outer */, tg_operator** out);

extern "C" tg_status tg_lookup_join_create(tg_session* s, tg_join_bridge* bridge,
    const int32_t* probe_types, int32_t n_probe_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* probe_output_channels, int32_t n_probe_output,
    tg_operator** out)
{
    if (!s || !bridge || !key_channels || n_key_channels < 1 || n_key_channels > 7) {
        TG_SET_ERR("1..7 join key channels");
        return TG_ERR_INVALID_ARG;
    }
    auto* op = new LookupJoinOp();
    op->s = s;
    op->bridge = bridge;
    op->key_channels.assign(key_channels, key_channels + n_key_channels);
    for (int i = 0; i < n_probe_channels; i++)
        op->probe_types.push_back((tg_type)probe_types[i]);
    op->probe_output.assign(probe_output_channels, probe_output_channels + n_probe_output);
    *out = op;
    return TG_OK;
}

extern "C" tg_status tg_lookup_join_create_ex(tg_session* s, tg_join_bridge* bridge,
    const int32_t* probe_types, int32_t n_probe_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* probe_output_channels, int32_t n_probe_output,
    int32_t join_type, tg_operator** out)
{
    tg_status st = tg_lookup_join_create(s, bridge, probe_types, n_probe_channels,
                                         key_channels, n_key_channels,
                                         probe_output_channels, n_probe_output, out);
    if (st != TG_OK) return st;
    static_cast<LookupJoinOp*>(*out)->outer = (join_type == 1);
    return TG_OK;
}

/* dynamic filter: the build side's non-null key range (valid after the
 * builder's finish); has_rows=0 means the probe side can skip entirely */
extern "C" tg_status tg_join_bridge_key_range(tg_join_bridge* b, int64_t* key_min,
                                              int64_t* key_max, int64_t* key_rows)
{
    if (!b || !b->built) { TG_SET_ERR("lookup source not built"); return TG_ERR_STATE; }
    if (b->t.generic) { TG_SET_ERR("key range is tracked for single BIGINT keys"); return TG_ERR_UNSUPPORTED; }
    if (key_min) *key_min = b->key_min;
    if (key_max) *key_max = b->key_max;
    if (key_rows) *key_rows = b->key_rows;
    return TG_OK;
}

/* request a probe-side dynamic-filter bitmap from this bridge's build
 * (must be called before the builder's finish) */
extern "C" tg_status tg_join_bridge_request_bitmap(tg_join_bridge* b)
{
    if (!b) { TG_SET_ERR("null bridge"); return TG_ERR_INVALID_ARG; }
    b->want_bitmap = true;
    return TG_OK;
}

/* internal: fetch the bitmap for the filter operator (ops_filter.hip) */
bool tg_bridge_df(tg_join_bridge* b, const uint64_t** bm, int64_t* mn, int64_t* mx)
{
    if (!b || !b->built || !b->t.set_bitmap) return false;
    *bm = b->t.set_bitmap;
    *mn = b->t.set_min;
    *mx = b->t.set_max;
    return true;
}
