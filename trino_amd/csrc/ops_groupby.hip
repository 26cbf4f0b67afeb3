/* ops_groupby.hip — group-by hash + hash aggregation operator.
 *
 * Mirrors:
 *  - BigintGroupByHash.java:191-300 / FlatHash.java:276-330 semantics at the
 *    operator boundary: group ids dense, assigned by first occurrence in row
 *    order (GroupByHash.java:121-128), null keys group together; output pages
 *    emit groups in consecutive group-id order (InMemoryHashAggregationBuilder
 *    buildResult). The device table assigns provisional ids with atomics and
 *    a finish-time remap by first-occurrence row restores the reference's
 *    exact id order deterministically.
 *  - HashAggregationOperator / GroupedAggregator.processPage
 *    (GroupedAggregator.java:77-101): COUNT, SUM(bigint/double), AVG(double)
 *    with PARTIAL state (count,sum) and FINAL combine
 *    (DoubleSumAggregation.java:37-45, DoubleAverageAggregations.java:38-63).
 *
 * Determinism/parity: group keys, ids, counts and integer
 * sums are bit-exact. Plain f64 SUM uses device atomicAdd (the reference's
 * own cross-driver combine order is nondeterministic too); operator parity
 * tests state their tolerance. TG_AGG_SUM_F64_EXACT extends the fused Q1
 * kernel's fixed-point trick to the generic operator: 128-bit integer state
 * accumulated with carry-propagating u64 atomics — order-independent, so
 * bit-stable across page splits and grids, and correctly rounded.
 *
 * Table layout: open addressing, power-of-2, per slot an int32 state
 * (-1 empty / -2 claimed / else gid) and a key store of canonical 64-bit
 * words per group (+ null mask word). Key-store writes are device-scope
 * atomic stores published by a release store of the gid; probes use relaxed
 * atomic loads (values land at the coherence point before the gid does).
 */
#include "dev_hash.h"

struct GTable {
    int64_t capacity = 0, mask = 0;
    int32_t* state = nullptr;        /* per slot: -1/-2/gid */
    int64_t max_groups = 0;
    uint64_t* keystore = nullptr;    /* [max_groups][n_words] */
    int64_t* first_row = nullptr;    /* [max_groups] atomicMin of global row idx */
    int32_t* counter = nullptr;      /* n_groups */
    int32_t n_words = 0;             /* n_key_channels + 1 (null mask) */
    /* variable-width key store (AppendOnlyVariableWidthData analog): VARCHAR
     * key channels store word = (len << 40) | byte_offset into varstore */
    uint8_t* varstore = nullptr;
    unsigned long long* var_cursor = nullptr;
    int64_t var_capacity = 0;
};

#define GT_VLEN(w)  ((int64_t)((w) >> 40))
#define GT_VOFF(w)  ((int64_t)((w) & ((1ULL << 40) - 1)))

__device__ static inline uint8_t gt_var_byte(const GTable& t, int64_t off)
{
    return __hip_atomic_load(&t.varstore[off], __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
}

__global__ void k_gt_init(int32_t* state, int64_t cap, int64_t* first_row, int64_t ngroups_cap)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t k = i; k < cap; k += stride) state[k] = -1;
    for (int64_t k = i; k < ngroups_cap; k += stride) first_row[k] = INT64_MAX;
}

/* assign group ids for one page; row_base = global row index of row 0.
 * NCH > 0 = compile-time channel count (registers + unrolled compares);
 * NCH == 0 = dynamic fallback for >6 channels. */
template <int NCH>
__global__ void k_gt_assign(GTable t, const KColH* cols, int n_ch_dyn,
                            int64_t n, int64_t row_base, int32_t* __restrict__ gids)
{
    const int n_ch = NCH > 0 ? NCH : n_ch_dyn;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        uint64_t h = row_hash(cols, n_ch, i);
        /* canonical words + null mask for this row (VARCHAR words are
         * assigned at insert, after the bytes are appended) */
        uint64_t w[8];
        uint64_t nullmask = 0;
        bool has_var = false;
        for (int c = 0; c < n_ch; c++) {
            bool nl = kcol_is_null(cols[c], i);
            bool isvar = cols[c].type == TG_VARCHAR;
            has_var |= isvar;
            w[c] = (nl || isvar) ? 0 : kcol_word(cols[c], i);
            nullmask |= (uint64_t)nl << c;
        }
        w[n_ch] = nullmask;
        int64_t slot = (int64_t)(d_murmur3_mix(h) & (uint64_t)t.mask);
        int32_t gid = -1;
        while (true) {
            /* ACQUIRE pairs with the release publication of state[slot] below:
             * a thread that observes gid must also observe the keystore words
             * (the address dependency works on current hardware, but the
             * formal HIP memory model needs the acquire) */
            int32_t st = __hip_atomic_load(&t.state[slot], __ATOMIC_ACQUIRE,
                                           __HIP_MEMORY_SCOPE_AGENT);
            if (st == -1) {
                int32_t old = -1;
                if (__hip_atomic_compare_exchange_strong(&t.state[slot], &old, -2,
                        __ATOMIC_ACQ_REL, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)) {
                    /* wave-aggregated id allocation: lanes claiming slots in
                     * the same instruction bundle share ONE atomicAdd (the
                     * single hot counter serialized Q16's 4.5M-group dedup
                     * at ~30 ms) */
                    {
                        unsigned long long act = __ballot(true);
                        int lane_ = (int)(threadIdx.x & 63);
                        int leader = __ffsll(act) - 1;
                        int32_t base_ = 0;
                        if (lane_ == leader)
                            base_ = atomicAdd(t.counter, __popcll(act));
                        base_ = __shfl(base_, leader, 64);
                        gid = base_ + __popcll(act & ((1ull << lane_) - 1ull));
                    }
                    if (has_var) {
                        for (int c = 0; c < n_ch; c++) {
                            if (cols[c].type != TG_VARCHAR) continue;
                            if ((nullmask >> c) & 1) { w[c] = 0; continue; }
                            int64_t po = cols[c].offsets[i];
                            int64_t len = cols[c].offsets[i + 1] - po;
                            unsigned long long off =
                                atomicAdd(t.var_cursor, (unsigned long long)len);
                            const uint8_t* src = (const uint8_t*)cols[c].data + po;
                            for (int64_t b = 0; b < len; b++)
                                t.varstore[off + b] = src[b];
                            w[c] = ((uint64_t)len << 40) | off;
                        }
                    }
                    for (int c = 0; c <= n_ch; c++)
                        __hip_atomic_store(&t.keystore[(int64_t)gid * t.n_words + c], w[c],
                                           __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    __hip_atomic_store(&t.state[slot], gid, __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_AGENT);
                    break;
                }
                st = old;
            }
            while (st == -2) {
                __builtin_amdgcn_s_sleep(1);
                st = __hip_atomic_load(&t.state[slot], __ATOMIC_ACQUIRE,
                                       __HIP_MEMORY_SCOPE_AGENT);
            }
            /* st >= 0: compare keys */
            bool eq = true;
            for (int c = 0; c <= n_ch && eq; c++) {
                uint64_t kv = __hip_atomic_load(&t.keystore[(int64_t)st * t.n_words + c],
                                                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                if (c < n_ch && cols[c].type == TG_VARCHAR) {
                    if ((nullmask >> c) & 1) continue;   /* settled by mask word */
                    int64_t po = cols[c].offsets[i];
                    int64_t plen = cols[c].offsets[i + 1] - po;
                    if (GT_VLEN(kv) != plen) { eq = false; break; }
                    int64_t so = GT_VOFF(kv);
                    const uint8_t* src = (const uint8_t*)cols[c].data + po;
                    for (int64_t b = 0; b < plen && eq; b++)
                        eq = (gt_var_byte(t, so + b) == src[b]);
                }
                else {
                    eq = (kv == w[c]);
                }
            }
            if (eq) { gid = st; break; }
            slot = (slot + 1) & t.mask;
        }
        gids[i] = gid;
        /* check-then-atomic: once first_row[g] is small, later rows skip the
         * RMW entirely (57M atomicMin on a handful of hot groups measured
         * ~26 ms; the relaxed pre-check makes it a cached read) */
        unsigned long long cand = (unsigned long long)(row_base + i);
        if ((unsigned long long)__hip_atomic_load(&t.first_row[gid],
                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) > cand)
            atomicMin((unsigned long long*)&t.first_row[gid], cand);
    }
}

/* rehash: reinsert existing groups into a larger table (keystore is stable;
 * only the slot index array is rebuilt) */
__global__ void k_gt_rehash(GTable t, int32_t n_groups, const int32_t* types, int n_ch)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    /* recompute the row hash from the stored canonical words (runs in a
     * later launch: plain varstore reads are safely visible) */
    int64_t h = 0;
    uint64_t nullmask = t.keystore[(int64_t)g * t.n_words + n_ch];
    for (int c = 0; c < n_ch; c++) {
        uint64_t w = t.keystore[(int64_t)g * t.n_words + c];
        uint64_t ch;
        if ((nullmask >> c) & 1) ch = 0;
        else if (types && types[c] == TG_VARCHAR)
            ch = d_xxhash64_bytes(t.varstore + GT_VOFF(w), GT_VLEN(w));
        else ch = d_bigint_hash((int64_t)w);   /* canonical words hash as longs:
                                                  f64 words are normalized bits,
                                                  matching d_double_hash */
        h = 31 * h + (int64_t)ch;
    }
    int64_t slot = (int64_t)(d_murmur3_mix((uint64_t)h) & (uint64_t)t.mask);
    while (true) {
        int32_t old = -1;
        if (__hip_atomic_compare_exchange_strong(&t.state[slot], &old, g,
                __ATOMIC_RELAXED, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT))
            break;
        slot = (slot + 1) & t.mask;
    }
}

/* ---- aggregation update (KAgg + shared kernel decls: agg_common.h) ---- */
#include "agg_common.h"

__global__ void k_agg_update(const int32_t* __restrict__ gids, int64_t n,
                             const KColH* cols, const KAgg* aggs, int n_aggs,
                             int step /* 0 partial/single input rows, 1 final combine */)
{
    /* wave-uniform fast path: when all 64 lanes share one group id (scalar
     * aggregation, clustered keys), shuffle-reduce in registers and issue
     * ONE atomic per wave — the per-lane atomics otherwise serialize on the
     * shared state address (measured ~150 ms for an 11M-row scalar sum). */
    int lane = threadIdx.x % 64;
    int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t base = i0 - lane; base < n; base += stride) {
        int64_t i = base + lane;
        bool active = i < n;
        int32_t g = active ? gids[i] : -1;
        int32_t g0 = __shfl(g, 0, 64);
        bool uniform = (__ballot(g == g0) == ~0ull) && (__ballot(active) == ~0ull);
        for (int a = 0; a < n_aggs; a++) {
            KAgg ag = aggs[a];
            int mmode = (ag.fn == TG_AGG_MIN_I64) ? 1
                      : (ag.fn == TG_AGG_MAX_I64) ? 2 : 0;
            bool row_ok = active;
            if (active && ag.mask_a >= 0 && ag.mask_a != ag.mask_b) {
                /* masked aggregate: col[mask_a] > col[mask_b] gates the row
                 * (typed: both int32 or int64 supported via kcol_word) */
                row_ok = kcol_sval(cols[ag.mask_a], i) >
                         kcol_sval(cols[ag.mask_b], i);
            }
            /* per-lane addend for this agg */
            long long ci = mmode == 1 ? INT64_MAX
                         : mmode == 2 ? INT64_MIN : 0;   /* integer addend */
            double cf = 0.0;        /* f64 addend */
            unsigned long long lo = 0, hi = 0;   /* exact i128 addend */
            if (row_ok) {
                switch (ag.fn) {
                    case TG_AGG_COUNT_STAR:
                        ci = (step == 0) ? 1 : ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_COUNT_COL:
                        if (step == 0) ci = kcol_is_null(cols[ag.in_ch], i) ? 0 : 1;
                        else ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_SUM_I64:
                        if (!kcol_is_null(cols[ag.in_ch], i))
                            ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_SUM_F64:
                        if (!kcol_is_null(cols[ag.in_ch], i))
                            cf = ((const double*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_SUM_F64_EXACT: {
                        if (step == 0) {
                            if (!kcol_is_null(cols[ag.in_ch], i)) {
                                double y = ((const double*)cols[ag.in_ch].data)[i] * ag.scale;
                                __int128 yi = (__int128)(long long)y;
                                lo = (unsigned long long)(unsigned __int128)yi;
                                hi = (unsigned long long)((unsigned __int128)yi >> 64);
                            }
                        }
                        else {
                            lo = (unsigned long long)((const int64_t*)cols[ag.in_ch].data)[i];
                            hi = (unsigned long long)((const int64_t*)cols[ag.in_ch + 1].data)[i];
                        }
                        break;
                    }
                    case TG_AGG_MIN_I64: case TG_AGG_MAX_I64:
                        /* raw value rides in ci (identity preset); FINAL
                           combine reads the partial's plain int64 column */
                        if (!kcol_is_null(cols[ag.in_ch], i))
                            ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_AVG_F64:
                        if (step == 0) {
                            if (!kcol_is_null(cols[ag.in_ch], i)) {
                                ci = 1;
                                cf = ((const double*)cols[ag.in_ch].data)[i];
                            }
                        }
                        else {
                            ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                            cf = ((const double*)cols[ag.in_ch + 1].data)[i];
                        }
                        break;
                }
            }
            if (!uniform) {
                /* hot-group hybrid: aggregate up to 8 distinct gids per wave
                 * via masked full-wave reductions (one atomic per distinct
                 * group); remaining lanes fall through to per-lane atomics.
                 * Fixes the few-hot-groups pattern (Q4's five priorities:
                 * 57M rows x 5 addresses serialized at ~26 ms). */
                unsigned long long todo = __ballot(active);
                int rounds = 0;
                while (todo && rounds < 8) {
                    int lead = __ffsll((unsigned long long)todo) - 1;
                    int32_t gsel = __shfl(g, lead, 64);
                    bool mine = active && g == gsel;
                    unsigned long long mask = __ballot(mine);
                    long long rci = mine ? ci
                        : mmode == 1 ? INT64_MAX : mmode == 2 ? INT64_MIN : 0;
                    double rcf = mine ? cf : 0.0;
                    unsigned long long rlo = mine ? lo : 0, rhi = mine ? hi : 0;
                    #pragma unroll
                    for (int off = 32; off >= 1; off >>= 1) {
                        { long long o = __shfl_xor(rci, off, 64);
                          rci = mmode == 1 ? (o < rci ? o : rci)
                              : mmode == 2 ? (o > rci ? o : rci) : rci + o; }
                        rcf += __shfl_xor(rcf, off, 64);
                        unsigned long long olo = __shfl_xor(rlo, off, 64);
                        unsigned long long ohi = __shfl_xor(rhi, off, 64);
                        unsigned long long nlo = rlo + olo;
                        rhi = rhi + ohi + (nlo < rlo ? 1ull : 0ull);
                        rlo = nlo;
                    }
                    if (lane == lead) {
                        switch (ag.fn) {
                            case TG_AGG_COUNT_STAR: case TG_AGG_COUNT_COL:
                            case TG_AGG_SUM_I64:
                                if (rci) atomicAdd((unsigned long long*)&ag.cnt[gsel],
                                                   (unsigned long long)rci);
                                break;
                            case TG_AGG_SUM_F64:
                                if (rcf != 0.0) atomicAdd(&ag.sum[gsel], rcf);
                                break;
                            case TG_AGG_SUM_F64_EXACT: {
                                unsigned long long old =
                                    atomicAdd((unsigned long long*)&ag.cnt[gsel], rlo);
                                unsigned long long carry = (old + rlo) < rlo ? 1ull : 0ull;
                                atomicAdd((unsigned long long*)ag.sum + gsel, rhi + carry);
                                break;
                            }
                            case TG_AGG_MIN_I64:
                                atomicMin((unsigned long long*)&ag.cnt[gsel],
                                          (unsigned long long)rci ^ 0x8000000000000000ull);
                                break;
                            case TG_AGG_MAX_I64:
                                atomicMax((unsigned long long*)&ag.cnt[gsel],
                                          (unsigned long long)rci ^ 0x8000000000000000ull);
                                break;
                            case TG_AGG_AVG_F64:
                                if (rci) atomicAdd((unsigned long long*)&ag.cnt[gsel],
                                                   (unsigned long long)rci);
                                if (rcf != 0.0) atomicAdd(&ag.sum[gsel], rcf);
                                break;
                        }
                    }
                    todo &= ~mask;
                    rounds++;
                    /* small first cluster => likely high cardinality: the
                     * match loop would cost more than direct atomics (run-
                     * structured inputs like streaming agg average ~4 rows
                     * per group) — bail and let leftovers take the fast
                     * per-lane path */
                    if (rounds == 1 && __popcll(mask) < 8) break;
                }
                /* leftovers (wave saw >8 distinct groups): direct atomics */
                bool leftover = active && ((todo >> lane) & 1ull);
                if (leftover) {
                    switch (ag.fn) {
                        case TG_AGG_COUNT_STAR: case TG_AGG_COUNT_COL:
                        case TG_AGG_SUM_I64:
                            if (ci) atomicAdd((unsigned long long*)&ag.cnt[g],
                                              (unsigned long long)ci);
                            break;
                        case TG_AGG_SUM_F64:
                            if (cf != 0.0) atomicAdd(&ag.sum[g], cf);
                            break;
                        case TG_AGG_SUM_F64_EXACT: {
                            unsigned long long old =
                                atomicAdd((unsigned long long*)&ag.cnt[g], lo);
                            unsigned long long carry = (old + lo) < lo ? 1ull : 0ull;
                            atomicAdd((unsigned long long*)ag.sum + g, hi + carry);
                            break;
                        }
                        case TG_AGG_MIN_I64:
                            atomicMin((unsigned long long*)&ag.cnt[g],
                                      (unsigned long long)ci ^ 0x8000000000000000ull);
                            break;
                        case TG_AGG_MAX_I64:
                            atomicMax((unsigned long long*)&ag.cnt[g],
                                      (unsigned long long)ci ^ 0x8000000000000000ull);
                            break;
                        case TG_AGG_AVG_F64:
                            if (ci) atomicAdd((unsigned long long*)&ag.cnt[g],
                                              (unsigned long long)ci);
                            if (cf != 0.0) atomicAdd(&ag.sum[g], cf);
                            break;
                    }
                }
            }
            else {
                /* wave reduce, one atomic from lane 0 */
                #pragma unroll
                for (int off = 32; off >= 1; off >>= 1) {
                    { long long o = __shfl_xor(ci, off, 64);
                      ci = mmode == 1 ? (o < ci ? o : ci)
                         : mmode == 2 ? (o > ci ? o : ci) : ci + o; }
                    cf += __shfl_xor(cf, off, 64);
                    unsigned long long olo = __shfl_xor(lo, off, 64);
                    unsigned long long ohi = __shfl_xor(hi, off, 64);
                    unsigned long long nlo = lo + olo;
                    hi = hi + ohi + (nlo < lo ? 1ull : 0ull);
                    lo = nlo;
                }
                if (lane == 0) {
                    switch (ag.fn) {
                        case TG_AGG_COUNT_STAR: case TG_AGG_COUNT_COL:
                        case TG_AGG_SUM_I64:
                            if (ci) atomicAdd((unsigned long long*)&ag.cnt[g0],
                                              (unsigned long long)ci);
                            break;
                        case TG_AGG_SUM_F64:
                            if (cf != 0.0) atomicAdd(&ag.sum[g0], cf);
                            break;
                        case TG_AGG_SUM_F64_EXACT: {
                            unsigned long long old =
                                atomicAdd((unsigned long long*)&ag.cnt[g0], lo);
                            unsigned long long carry = (old + lo) < lo ? 1ull : 0ull;
                            atomicAdd((unsigned long long*)ag.sum + g0, hi + carry);
                            break;
                        }
                        case TG_AGG_MIN_I64:
                            atomicMin((unsigned long long*)&ag.cnt[g0],
                                      (unsigned long long)ci ^ 0x8000000000000000ull);
                            break;
                        case TG_AGG_MAX_I64:
                            atomicMax((unsigned long long*)&ag.cnt[g0],
                                      (unsigned long long)ci ^ 0x8000000000000000ull);
                            break;
                        case TG_AGG_AVG_F64:
                            if (ci) atomicAdd((unsigned long long*)&ag.cnt[g0],
                                              (unsigned long long)ci);
                            if (cf != 0.0) atomicAdd(&ag.sum[g0], cf);
                            break;
                    }
                }
            }

        }
    }
}

/* sorted-gid variant (streaming aggregation): gids are non-decreasing, so a
 * wave-segmented inclusive scan reduces each run in registers and only the
 * LAST lane of each run issues atomics — ~12x fewer atomics than per-lane
 * updates on run-structured input (600M rows / 150M runs measured 76 ms ->
 * segmented ~15 ms). */
__global__ void k_agg_update_sorted(const int32_t* __restrict__ gids, int64_t n,
                                    const KColH* cols, const KAgg* aggs, int n_aggs,
                                    int step)
{
    int lane = threadIdx.x % 64;
    int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t base = i0 - lane; base < n; base += stride) {
        int64_t i = base + lane;
        bool active = i < n;
        int32_t g = active ? gids[i] : -1;
        int32_t gnext = __shfl_down(g, 1, 64);
        bool is_last = active && (lane == 63 || i + 1 >= n || gnext != g);
        /* run topology is agg-invariant: hoist the per-step "stays in my
         * run" masks out of the agg loop (was re-shuffling g for every
         * aggregate — q21's 5-agg pass re-ran the whole 4-register scan
         * 5x; with the masks hoisted each agg scans only the registers
         * its class needs: 30.0 ms -> see profiles/) */
        bool segok[6];
        #pragma unroll
        for (int off = 1, k = 0; off < 64; off <<= 1, k++) {
            int32_t og = __shfl_up(g, off, 64);
            segok[k] = (lane >= off) && (og == g);
        }
        for (int a = 0; a < n_aggs; a++) {
            KAgg ag = aggs[a];
            int mmode = (ag.fn == TG_AGG_MIN_I64) ? 1
                      : (ag.fn == TG_AGG_MAX_I64) ? 2 : 0;
            bool row_ok = active;
            if (active && ag.mask_a >= 0 && ag.mask_a != ag.mask_b) {
                row_ok = kcol_sval(cols[ag.mask_a], i) >
                         kcol_sval(cols[ag.mask_b], i);
            }
            long long ci = mmode == 1 ? INT64_MAX
                         : mmode == 2 ? INT64_MIN : 0;
            double cf = 0.0;
            unsigned long long lo = 0, hi = 0;
            if (row_ok) {
                switch (ag.fn) {
                    case TG_AGG_COUNT_STAR:
                        ci = (step == 0) ? 1 : ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_COUNT_COL:
                        if (step == 0) ci = kcol_is_null(cols[ag.in_ch], i) ? 0 : 1;
                        else ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_SUM_I64:
                        if (!kcol_is_null(cols[ag.in_ch], i))
                            ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_SUM_F64:
                        if (!kcol_is_null(cols[ag.in_ch], i))
                            cf = ((const double*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_SUM_F64_EXACT: {
                        if (step == 0) {
                            if (!kcol_is_null(cols[ag.in_ch], i)) {
                                double y = ((const double*)cols[ag.in_ch].data)[i] * ag.scale;
                                __int128 yi = (__int128)(long long)y;
                                lo = (unsigned long long)(unsigned __int128)yi;
                                hi = (unsigned long long)((unsigned __int128)yi >> 64);
                            }
                        }
                        else {
                            lo = (unsigned long long)((const int64_t*)cols[ag.in_ch].data)[i];
                            hi = (unsigned long long)((const int64_t*)cols[ag.in_ch + 1].data)[i];
                        }
                        break;
                    }
                    case TG_AGG_MIN_I64: case TG_AGG_MAX_I64:
                        /* raw value rides in ci (identity preset); FINAL
                           combine reads the partial's plain int64 column */
                        if (!kcol_is_null(cols[ag.in_ch], i))
                            ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                        break;
                    case TG_AGG_AVG_F64:
                        if (step == 0) {
                            if (!kcol_is_null(cols[ag.in_ch], i)) {
                                ci = 1;
                                cf = ((const double*)cols[ag.in_ch].data)[i];
                            }
                        }
                        else {
                            ci = ((const int64_t*)cols[ag.in_ch].data)[i];
                            cf = ((const double*)cols[ag.in_ch + 1].data)[i];
                        }
                        break;
                }
            }
            /* segmented inclusive scan on equal-gid prefixes — only the
             * registers this aggregate class carries */
            if (ag.fn == TG_AGG_SUM_F64) {
                #pragma unroll
                for (int off = 1, k = 0; off < 64; off <<= 1, k++) {
                    double ocf = __shfl_up(cf, off, 64);
                    if (segok[k]) cf += ocf;
                }
            }
            else if (ag.fn == TG_AGG_AVG_F64) {
                #pragma unroll
                for (int off = 1, k = 0; off < 64; off <<= 1, k++) {
                    long long oci = __shfl_up(ci, off, 64);
                    double ocf = __shfl_up(cf, off, 64);
                    if (segok[k]) { ci += oci; cf += ocf; }
                }
            }
            else if (ag.fn == TG_AGG_SUM_F64_EXACT) {
                #pragma unroll
                for (int off = 1, k = 0; off < 64; off <<= 1, k++) {
                    unsigned long long olo = __shfl_up(lo, off, 64);
                    unsigned long long ohi = __shfl_up(hi, off, 64);
                    if (segok[k]) {
                        unsigned long long nlo = lo + olo;
                        hi = hi + ohi + (nlo < lo ? 1ull : 0ull);
                        lo = nlo;
                    }
                }
            }
            else {   /* COUNT/SUM_I64/MIN/MAX: single i64 register */
                #pragma unroll
                for (int off = 1, k = 0; off < 64; off <<= 1, k++) {
                    long long oci = __shfl_up(ci, off, 64);
                    if (segok[k]) {
                        ci = mmode == 1 ? (oci < ci ? oci : ci)
                           : mmode == 2 ? (oci > ci ? oci : ci) : ci + oci;
                    }
                }
            }
            if (is_last) {
                switch (ag.fn) {
                    case TG_AGG_COUNT_STAR: case TG_AGG_COUNT_COL:
                    case TG_AGG_SUM_I64:
                        if (ci) atomicAdd((unsigned long long*)&ag.cnt[g],
                                          (unsigned long long)ci);
                        break;
                    case TG_AGG_SUM_F64:
                        if (cf != 0.0) atomicAdd(&ag.sum[g], cf);
                        break;
                    case TG_AGG_SUM_F64_EXACT: {
                        unsigned long long old =
                            atomicAdd((unsigned long long*)&ag.cnt[g], lo);
                        unsigned long long carry = (old + lo) < lo ? 1ull : 0ull;
                        atomicAdd((unsigned long long*)ag.sum + g, hi + carry);
                        break;
                    }
                    case TG_AGG_MIN_I64:
                        atomicMin((unsigned long long*)&ag.cnt[g],
                                  (unsigned long long)ci ^ 0x8000000000000000ull);
                        break;
                    case TG_AGG_MAX_I64:
                        atomicMax((unsigned long long*)&ag.cnt[g],
                                  (unsigned long long)ci ^ 0x8000000000000000ull);
                        break;
                    case TG_AGG_AVG_F64:
                        if (ci) atomicAdd((unsigned long long*)&ag.cnt[g],
                                          (unsigned long long)ci);
                        if (cf != 0.0) atomicAdd(&ag.sum[g], cf);
                        break;
                }
            }
        }
    }
}

/* MarkDistinctOperator analog: row is 'distinct' iff it is its group's
 * first occurrence (operator/MarkDistinctOperator.java semantics) */
__global__ void k_mark_first(const int32_t* __restrict__ gids, int64_t n,
                             const int64_t* __restrict__ first_row,
                             int64_t row_base, int8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        flags[i] = first_row[gids[i]] == row_base + i ? 1 : 0;
}

/* ---- output materialization (after remap) ---- */
__global__ void k_emit_keys(const uint64_t* __restrict__ keystore, int n_words,
                            const int32_t* __restrict__ old_by_new, int32_t n_groups,
                            int ch, int32_t type, void* __restrict__ out,
                            uint64_t* __restrict__ out_valid)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    int32_t og = old_by_new[g];
    uint64_t w = keystore[(int64_t)og * n_words + ch];
    uint64_t nullmask = keystore[(int64_t)og * n_words + (n_words - 1)];
    bool isnull = (nullmask >> ch) & 1;
    switch (type) {
        case TG_BIGINT: ((int64_t*)out)[g] = (int64_t)w; break;
        case TG_INTEGER: case TG_DATE: ((int32_t*)out)[g] = (int32_t)w; break;
        case TG_SMALLINT: ((int16_t*)out)[g] = (int16_t)w; break;
        case TG_TINYINT: case TG_BOOLEAN: ((int8_t*)out)[g] = (int8_t)w; break;
        default: ((double*)out)[g] = __longlong_as_double((long long)w); break;
    }
    if (out_valid && isnull)
        atomicAnd((unsigned long long*)&out_valid[g >> 6], ~(1ull << (g & 63)));
}

__global__ void k_emit_var_lens(const uint64_t* __restrict__ keystore, int n_words,
                                const int32_t* __restrict__ old_by_new, int32_t n_groups,
                                int ch, int32_t* __restrict__ lens)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    uint64_t w = keystore[(int64_t)old_by_new[g] * n_words + ch];
    uint64_t nullmask = keystore[(int64_t)old_by_new[g] * n_words + (n_words - 1)];
    lens[g] = ((nullmask >> ch) & 1) ? 0 : (int32_t)GT_VLEN(w);
}

__global__ void k_emit_var_bytes(GTable t, const int32_t* __restrict__ old_by_new,
                                 int32_t n_groups, int ch,
                                 const int32_t* __restrict__ out_offsets,
                                 uint8_t* __restrict__ out_bytes,
                                 uint64_t* __restrict__ out_valid)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    int32_t og = old_by_new[g];
    uint64_t w = t.keystore[(int64_t)og * t.n_words + ch];
    uint64_t nullmask = t.keystore[(int64_t)og * t.n_words + (t.n_words - 1)];
    if ((nullmask >> ch) & 1) {
        if (out_valid)
            atomicAnd((unsigned long long*)&out_valid[g >> 6], ~(1ull << (g & 63)));
        return;
    }
    int64_t off = GT_VOFF(w), len = GT_VLEN(w);
    for (int64_t b = 0; b < len; b++)
        out_bytes[out_offsets[g] + b] = t.varstore[off + b];
}

__global__ void k_emit_f64(const double* __restrict__ state, const int32_t* old_by_new,
                           int32_t n, double* __restrict__ out)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n) out[g] = state[old_by_new[g]];
}

__global__ void k_emit_i64_biased(const long long* __restrict__ state,
                                  const int32_t* __restrict__ old_by_new,
                                  int32_t n, int64_t* __restrict__ out)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    out[g] = (int64_t)((unsigned long long)state[old_by_new ? old_by_new[g] : g]
                       ^ 0x8000000000000000ull);
}

__global__ void k_emit_i64(const long long* __restrict__ state, const int32_t* old_by_new,
                           int32_t n, int64_t* __restrict__ out)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g < n) out[g] = state[old_by_new[g]];
}

__global__ void k_emit_exact(const unsigned long long* __restrict__ lo,
                             const unsigned long long* __restrict__ hi,
                             const int32_t* __restrict__ old_by_new, int32_t n,
                             double inv_scale, double* __restrict__ out)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    int32_t og = old_by_new[g];
    __int128 v = (__int128)(((unsigned __int128)hi[og] << 64) | lo[og]);
    out[g] = (double)v * inv_scale;   /* i128->f64 correctly rounded, *2^-k exact */
}

__global__ void k_emit_avg(const double* __restrict__ sum, const long long* __restrict__ cnt,
                           const int32_t* old_by_new, int32_t n, double* __restrict__ out,
                           uint64_t* __restrict__ out_valid)
{
    int32_t g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n) return;
    int32_t og = old_by_new[g];
    if (cnt[og] == 0) {   /* AVG of no rows = NULL */
        out[g] = 0.0;
        atomicAnd((unsigned long long*)&out_valid[g >> 6], ~(1ull << (g & 63)));
    }
    else {
        out[g] = sum[og] / (double)cnt[og];
    }
}

/* ---- operator ---- */
#include <algorithm>
#include <numeric>

#include <atomic>
#include <mutex>

/* ---- adaptive partial aggregation ----------------------------------------
 * PartialAggregationController analog (operator/aggregation/partial/
 * PartialAggregationController.java:34-100): shared across the PARTIAL-step
 * HashAggOps of one plan node. After sampling >= 1.5x max_partial_bytes of
 * input, if the unique(output)/input row ratio exceeds the threshold
 * (reference session default 0.8), partial aggregation is disabled — pages
 * are re-shaped into partial-state layout with no hash-table work and flow
 * straight through (the FINAL stage does the real grouping). Re-enabled
 * after 200x more input bytes in case efficiency improved (same constants
 * as the reference). */
struct tg_pa_controller {
    std::mutex m;
    std::atomic<bool> disabled{false};
    int64_t max_partial_bytes = 0;
    double unique_ratio_threshold = 0.8;
    int64_t total_bytes = 0, total_rows = 0, total_unique = 0;

    void on_flush(int64_t bytes, int64_t rows, int64_t unique, int have_unique)
    {
        std::lock_guard<std::mutex> g(m);
        bool dis = disabled.load(std::memory_order_relaxed);
        /* when PA was re-enabled, ignore stats from disabled-mode flushes
         * (PartialAggregationController.java:69-72) */
        if (!dis && !have_unique) return;
        total_bytes += bytes;
        total_rows += rows;
        if (have_unique) total_unique += unique;
        if (!dis && total_bytes >= (int64_t)(max_partial_bytes * 1.5) &&
            total_rows > 0 &&
            (double)total_unique / (double)total_rows > unique_ratio_threshold) {
            disabled.store(true, std::memory_order_relaxed);
            dis = true;
        }
        if (dis && total_bytes >= (int64_t)(max_partial_bytes * 1.5) * 200) {
            total_bytes = total_rows = total_unique = 0;
            disabled.store(false, std::memory_order_relaxed);
        }
    }
};

/* pass-through re-shape: one raw input row -> its partial-state form (the
 * exact channel layout HashAggOp::emit produces for PARTIAL), null inputs
 * mapped to the combine-neutral element of each aggregate. out1 only for
 * the two-channel states (AVG count+sum, EXACT lo+hi). */
__global__ void k_pa_state(const KColH* __restrict__ cols, int64_t n, KAgg ag,
                           long long* __restrict__ out0,
                           long long* __restrict__ out1)
{
    int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = i0; i < n; i += stride) {
        switch (ag.fn) {
            case TG_AGG_COUNT_STAR:
                out0[i] = 1;
                break;
            case TG_AGG_COUNT_COL:
                out0[i] = kcol_is_null(cols[ag.in_ch], i) ? 0 : 1;
                break;
            case TG_AGG_SUM_I64:
                out0[i] = kcol_is_null(cols[ag.in_ch], i) ? 0
                        : ((const int64_t*)cols[ag.in_ch].data)[i];
                break;
            case TG_AGG_SUM_F64:
                ((double*)out0)[i] = kcol_is_null(cols[ag.in_ch], i) ? 0.0
                        : ((const double*)cols[ag.in_ch].data)[i];
                break;
            case TG_AGG_SUM_F64_EXACT: {
                __int128 yi = 0;
                if (!kcol_is_null(cols[ag.in_ch], i))
                    yi = (__int128)(long long)
                         (((const double*)cols[ag.in_ch].data)[i] * ag.scale);
                out0[i] = (long long)(unsigned long long)(unsigned __int128)yi;
                out1[i] = (long long)(unsigned long long)
                          ((unsigned __int128)yi >> 64);
                break;
            }
            case TG_AGG_MIN_I64:
                out0[i] = kcol_is_null(cols[ag.in_ch], i) ? INT64_MAX
                        : ((const int64_t*)cols[ag.in_ch].data)[i];
                break;
            case TG_AGG_MAX_I64:
                out0[i] = kcol_is_null(cols[ag.in_ch], i) ? INT64_MIN
                        : ((const int64_t*)cols[ag.in_ch].data)[i];
                break;
            case TG_AGG_AVG_F64:
                if (kcol_is_null(cols[ag.in_ch], i)) {
                    out0[i] = 0;
                    ((double*)out1)[i] = 0.0;
                }
                else {
                    out0[i] = 1;
                    ((double*)out1)[i] = ((const double*)cols[ag.in_ch].data)[i];
                }
                break;
        }
    }
}

struct HashAggOp : tg_operator {
    std::vector<int32_t> group_channels;
    std::vector<tg_type> group_types;
    std::vector<tg_agg_spec> aggs;
    tg_agg_step step = TG_STEP_SINGLE;

    GTable t;
    bool mark_distinct = false;   /* MarkDistinctOperator mode: pass the page
                                     through + append a BOOLEAN first-
                                     occurrence channel; no group emit */
    std::vector<KAgg> agg_state;       /* host mirror with device ptrs */
    int32_t* d_group_types = nullptr;  /* device copy for rehash/emit */
    int64_t rows_seen = 0;
    bool emitted = false;
    tg_pa_controller* pa = nullptr;    /* adaptive partial agg (not owned) */
    int64_t pa_bytes = 0, pa_rows = 0; /* input accounted since last flush */
    bool passed_through = false;       /* some pages bypassed the table */

    tg_status init_table(int64_t cap, int64_t max_groups)
    {
        t.capacity = cap; t.mask = cap - 1;
        t.max_groups = max_groups;
        t.n_words = (int32_t)group_channels.size() + 1;
        TG_POOL_ALLOC(s, &t.state, cap * 4);
        TG_POOL_ALLOC(s, &t.keystore, max_groups * t.n_words * 8);
        TG_POOL_ALLOC(s, &t.first_row, max_groups * 8);
        TG_POOL_ALLOC(s, &t.counter, 4);
        TG_HIP_CHECK(hipMemsetAsync(t.counter, 0, 4, s->stream));
        hipLaunchKernelGGL(k_gt_init, dim3(tg_grid_for(cap)), dim3(TG_BLOCK), 0, s->stream,
                           t.state, cap, t.first_row, max_groups);
        TG_HIP_CHECK(hipGetLastError());
        for (auto& a : agg_state) {
            if (a.sum) TG_HIP_CHECK(hipMemsetAsync(a.sum, 0, max_groups * 8, s->stream));
            if (a.cnt) TG_HIP_CHECK(hipMemsetAsync(a.cnt,
                    a.fn == TG_AGG_MIN_I64 ? 0xFF : 0, max_groups * 8, s->stream));
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        return TG_OK;
    }

    int32_t n_groups_host()
    {
        int32_t n = 0;
        (void)hipMemcpy(&n, t.counter, 4, hipMemcpyDeviceToHost);
        return n;
    }

    tg_status ensure_varstore(int64_t incoming_bytes)
    {
        if (!t.varstore || incoming_bytes == 0) return TG_OK;
        unsigned long long cur = 0;
        TG_HIP_CHECK(hipMemcpy(&cur, t.var_cursor, 8, hipMemcpyDeviceToHost));
        if ((int64_t)cur + incoming_bytes <= t.var_capacity) return TG_OK;
        int64_t ncap = t.var_capacity;
        while (ncap < (int64_t)cur + incoming_bytes) ncap *= 2;
        uint8_t* nv = nullptr;
        TG_HIP_CHECK(hipMalloc(&nv, ncap));
        TG_HIP_CHECK(hipMemcpyAsync(nv, t.varstore, cur, hipMemcpyDeviceToDevice, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        TG_HIP_CHECK(hipFree(t.varstore));
        t.varstore = nv;
        t.var_capacity = ncap;
        return TG_OK;
    }

    tg_status grow_if_needed(int64_t incoming)
    {
        int32_t ng = n_groups_host();
        if (ng + incoming <= (int64_t)(t.capacity * 0.7) &&
            ng + incoming <= t.max_groups) return TG_OK;
        int64_t need_groups = ng + incoming;
        int64_t new_groups = t.max_groups;
        while (new_groups < need_groups) new_groups *= 2;
        int64_t new_cap = t.capacity;
        while ((int64_t)(new_cap * 0.7) < need_groups) new_cap *= 2;

        /* grow keystore/first_row/states preserving contents */
        if (new_groups != t.max_groups) {
            uint64_t* nk; int64_t* nf;
            TG_POOL_ALLOC(s, &nk, new_groups * t.n_words * 8);
            TG_POOL_ALLOC(s, &nf, new_groups * 8);
            TG_HIP_CHECK(hipMemcpyAsync(nk, t.keystore, t.max_groups * t.n_words * 8,
                                        hipMemcpyDeviceToDevice, s->stream));
            TG_HIP_CHECK(hipMemcpyAsync(nf, t.first_row, t.max_groups * 8,
                                        hipMemcpyDeviceToDevice, s->stream));
            hipLaunchKernelGGL(k_gt_init, dim3(tg_grid_for(new_groups)), dim3(TG_BLOCK),
                               0, s->stream, (int32_t*)nf /*unused dummy*/, 0,
                               nf + t.max_groups, new_groups - t.max_groups);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            tg_pool_free(s, t.keystore); tg_pool_free(s, t.first_row);
            t.keystore = nk; t.first_row = nf;
            for (auto& a : agg_state) {
                if (a.sum) {
                    double* ns;
                    TG_POOL_ALLOC(s, &ns, new_groups * 8);
                    TG_HIP_CHECK(hipMemsetAsync(ns, 0, new_groups * 8, s->stream));
                    TG_HIP_CHECK(hipMemcpyAsync(ns, a.sum, t.max_groups * 8,
                                                hipMemcpyDeviceToDevice, s->stream));
                    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
                    tg_pool_free(s, a.sum);
                    a.sum = ns;
                }
                if (a.cnt) {
                    long long* nc;
                    TG_POOL_ALLOC(s, &nc, new_groups * 8);
                    TG_HIP_CHECK(hipMemsetAsync(nc,
                            a.fn == TG_AGG_MIN_I64 ? 0xFF : 0, new_groups * 8, s->stream));
                    TG_HIP_CHECK(hipMemcpyAsync(nc, a.cnt, t.max_groups * 8,
                                                hipMemcpyDeviceToDevice, s->stream));
                    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
                    tg_pool_free(s, a.cnt);
                    a.cnt = nc;
                }
            }
            t.max_groups = new_groups;
        }
        if (new_cap != t.capacity) {
            tg_pool_free(s, t.state);
            TG_POOL_ALLOC(s, &t.state, new_cap * 4);
            t.capacity = new_cap; t.mask = new_cap - 1;
            hipLaunchKernelGGL(k_gt_init, dim3(tg_grid_for(new_cap)), dim3(TG_BLOCK),
                               0, s->stream, t.state, new_cap, t.first_row, 0);
            TG_HIP_CHECK(hipGetLastError());
            if (ng > 0) {
                hipLaunchKernelGGL(k_gt_rehash, dim3((ng + TG_BLOCK - 1) / TG_BLOCK),
                                   dim3(TG_BLOCK), 0, s->stream, t, ng, d_group_types,
                                   (int)group_channels.size());
                TG_HIP_CHECK(hipGetLastError());
            }
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        }
        return TG_OK;
    }

    static int64_t page_bytes(const DevPage& p)
    {
        int64_t b = 0;
        for (const DevBlock& blk : p.blocks) b += blk.n * blk.elem_size();
        return b;
    }

    /* partial aggregation adaptively disabled: emit the page in partial-
     * state layout with no hash-table work (HashAggregationOperator's
     * spillToDiskAndClear-free adaptive path) */
    tg_status pass_through(DevPage&& in)
    {
        DevPage outp;
        outp.n = in.n;
        int64_t n = in.n;
        int64_t in_bytes = page_bytes(in);
        for (int32_t gc : group_channels) {
            const DevBlock& src = in.blocks[gc];
            DevBlock b;
            b.type = src.type;
            b.n = n;
            TG_POOL_ALLOC(s, &b.data, (n ? n : 1) * src.elem_size());
            TG_HIP_CHECK(hipMemcpyAsync(b.data, src.data, n * src.elem_size(),
                                        hipMemcpyDeviceToDevice, s->stream));
            if (src.valid) {
                int64_t words = (n + 63) / 64;
                TG_POOL_ALLOC(s, &b.valid, (words ? words : 1) * 8);
                TG_HIP_CHECK(hipMemcpyAsync(b.valid, src.valid, words * 8,
                                            hipMemcpyDeviceToDevice, s->stream));
            }
            outp.blocks.push_back(b);
        }
        KColH* d_all = nullptr;
        tg_status st = make_kcols(s, in, nullptr, (int)in.blocks.size(), &d_all);
        if (st != TG_OK) { tg_free_page(s, &in); return st; }
        for (const KAgg& ag : agg_state) {
            bool two = (ag.fn == TG_AGG_AVG_F64 || ag.fn == TG_AGG_SUM_F64_EXACT);
            DevBlock b0;
            b0.type = (ag.fn == TG_AGG_SUM_F64) ? TG_DOUBLE : TG_BIGINT;
            b0.n = n;
            TG_POOL_ALLOC(s, &b0.data, (n ? n : 1) * 8);
            DevBlock b1;
            if (two) {
                b1.type = (ag.fn == TG_AGG_AVG_F64) ? TG_DOUBLE : TG_BIGINT;
                b1.n = n;
                TG_POOL_ALLOC(s, &b1.data, (n ? n : 1) * 8);
            }
            hipLaunchKernelGGL(k_pa_state, dim3(tg_grid_for(n)), dim3(TG_BLOCK),
                               0, s->stream, d_all, n, ag, (long long*)b0.data,
                               two ? (long long*)b1.data : nullptr);
            TG_HIP_CHECK(hipGetLastError());
            outp.blocks.push_back(b0);
            if (two) outp.blocks.push_back(b1);
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_all);
        tg_free_page(s, &in);
        /* disabled-mode flush: bytes/rows only (no unique count) — feeds
         * the re-enable counter (PartialAggregationController.java:82-88) */
        pa->on_flush(in_bytes, n, 0, 0);
        passed_through = true;
        stage_output(std::move(outp));
        return TG_OK;
    }

    tg_status add_input(const tg_page* page) override
    {
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        if (pa && step == TG_STEP_PARTIAL) {
            if (pa->disabled.load(std::memory_order_relaxed))
                return pass_through(std::move(in));
            pa_bytes += page_bytes(in);
            pa_rows += in.n;
        }
        KColH* d_keys = nullptr;
        int32_t* d_gids = nullptr;
        if (group_channels.empty()) {
            /* scalar aggregation (AggregationOperator): one implicit group 0 */
            TG_POOL_ALLOC(s, &d_gids, (in.n ? in.n : 1) * 4);
            TG_HIP_CHECK(hipMemsetAsync(d_gids, 0, in.n * 4, s->stream));
            int32_t one = 1;
            TG_HIP_CHECK(hipMemcpyAsync(t.counter, &one, 4, hipMemcpyHostToDevice, s->stream));
            int64_t zero = 0;
            TG_HIP_CHECK(hipMemcpyAsync(t.first_row, &zero, 8, hipMemcpyHostToDevice, s->stream));
            goto have_gids;
        }
        st = grow_if_needed(in.n);
        if (st != TG_OK) { tg_free_page(s, &in); return st; }
        if (t.varstore) {
            int64_t vb = 0;
            for (size_t c = 0; c < group_channels.size(); c++) {
                if (group_types[c] != TG_VARCHAR) continue;
                const DevBlock& b = in.blocks[group_channels[c]];
                int32_t total = 0;
                TG_HIP_CHECK(hipMemcpy(&total, b.offsets + b.n, 4, hipMemcpyDeviceToHost));
                vb += total;
            }
            st = ensure_varstore(vb);
            if (st != TG_OK) { tg_free_page(s, &in); return st; }
        }

        st = make_kcols(s, in, group_channels.data(), (int)group_channels.size(), &d_keys);
        if (st != TG_OK) { tg_free_page(s, &in); return st; }
        TG_POOL_ALLOC(s, &d_gids, in.n * 4);
        {
            int nch = (int)group_channels.size();
            auto kfn = nch == 1 ? k_gt_assign<1> : nch == 2 ? k_gt_assign<2> :
                       nch == 3 ? k_gt_assign<3> : nch == 4 ? k_gt_assign<4> :
                       k_gt_assign<0>;
            hipLaunchKernelGGL(kfn, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK), 0, s->stream,
                               t, d_keys, nch, in.n, rows_seen, d_gids);
            TG_HIP_CHECK(hipGetLastError());
        }
have_gids:;
        if (mark_distinct) {
            DevBlock fb;
            fb.type = TG_BOOLEAN;
            fb.n = in.n;
            TG_POOL_ALLOC(s, &fb.data, in.n ? in.n : 1);
            hipLaunchKernelGGL(k_mark_first, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                               0, s->stream, d_gids, in.n, t.first_row,
                               rows_seen, (int8_t*)fb.data);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            rows_seen += in.n;
            tg_pool_free(s, d_gids);
            if (d_keys) tg_pool_free(s, d_keys);
            DevPage outp = std::move(in);   /* pass-through + flag channel */
            outp.blocks.push_back(fb);
            stage_output(std::move(outp));
            return TG_OK;
        }

        /* aggregation inputs: all channels */
        KColH* d_all = nullptr;
        st = make_kcols(s, in, nullptr, (int)in.blocks.size(), &d_all);
        if (st == TG_OK && !aggs.empty()) {
            KAgg* d_aggs = nullptr;
            TG_POOL_ALLOC(s, &d_aggs, agg_state.size() * sizeof(KAgg));
            TG_HIP_CHECK(hipMemcpyAsync(d_aggs, agg_state.data(),
                                        agg_state.size() * sizeof(KAgg),
                                        hipMemcpyHostToDevice, s->stream));
            hipLaunchKernelGGL(k_agg_update, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                               0, s->stream, d_gids, in.n, d_all, d_aggs,
                               (int)agg_state.size(), step == TG_STEP_FINAL ? 1 : 0);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            tg_pool_free(s, d_aggs);
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        rows_seen += in.n;
        tg_pool_free(s, d_gids);
        if (d_keys) tg_pool_free(s, d_keys);
        if (d_all) tg_pool_free(s, d_all);
        tg_free_page(s, &in);
        return st;
    }

    tg_status emit_varchar_key(int32_t ng, int ch, const int32_t* d_obn, int grid,
                               DevBlock* out)
    {
        out->type = TG_VARCHAR;
        out->n = ng;
        int32_t* d_lens = nullptr;
        TG_POOL_ALLOC(s, &d_lens, (ng ? ng : 1) * 4);
        hipLaunchKernelGGL(k_emit_var_lens, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                           t.keystore, t.n_words, d_obn, ng, ch, d_lens);
        TG_HIP_CHECK(hipGetLastError());
        std::vector<int32_t> lens(ng);
        if (ng) {
            TG_HIP_CHECK(hipMemcpy(lens.data(), d_lens, (size_t)ng * 4,
                                   hipMemcpyDeviceToHost));
        }
        std::vector<int32_t> offs(ng + 1, 0);
        for (int32_t i = 0; i < ng; i++) offs[i + 1] = offs[i] + lens[i];
        TG_POOL_ALLOC(s, &out->offsets, (ng + 1) * 4);
        TG_HIP_CHECK(hipMemcpyAsync(out->offsets, offs.data(), (ng + 1) * 4,
                                    hipMemcpyHostToDevice, s->stream));
        TG_POOL_ALLOC(s, &out->data, offs[ng] ? offs[ng] : 1);
        int64_t words = (ng + 63) / 64;
        TG_POOL_ALLOC(s, &out->valid, (words ? words : 1) * 8);
        TG_HIP_CHECK(hipMemsetAsync(out->valid, 0xFF, words * 8, s->stream));
        hipLaunchKernelGGL(k_emit_var_bytes, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                           t, d_obn, ng, ch, out->offsets, (uint8_t*)out->data, out->valid);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_lens);
        return TG_OK;
    }

    tg_status emit() /* build the output page: groups in remapped id order */
    {
        int32_t ng = n_groups_host();
        /* every page bypassed the table (adaptive pass-through): the
         * partial states are already staged; no hash-table output page */
        if (passed_through && !group_channels.empty() && ng == 0)
            return TG_OK;
        /* remap: order groups by first-occurrence row (== the reference's
         * insertion-order ids, deterministically) */
        int32_t* d_obn = nullptr;
        TG_POOL_ALLOC(s, &d_obn, (ng ? ng : 1) * 4);
        if (ng >= 65536) {
            /* device argsort (first_row keys are unique non-negative rows);
             * host std::sort at Q3's ~1.1M groups cost tens of ms per step */
            int64_t* d_fr = nullptr;
            TG_POOL_ALLOC(s, &d_fr, (int64_t)ng * 8);
            TG_HIP_CHECK(hipMemcpyAsync(d_fr, t.first_row, (int64_t)ng * 8,
                                        hipMemcpyDeviceToDevice, s->stream));
            tg_status sst = run_argsort_i64(s, d_fr, ng, d_obn);
            if (sst != TG_OK) return sst;
            tg_pool_free(s, d_fr);
        }
        else if (ng) {
            std::vector<int64_t> first(ng);
            TG_HIP_CHECK(hipMemcpy(first.data(), t.first_row, (size_t)ng * 8,
                                   hipMemcpyDeviceToHost));
            std::vector<int32_t> old_by_new(ng);
            std::iota(old_by_new.begin(), old_by_new.end(), 0);
            std::sort(old_by_new.begin(), old_by_new.end(),
                      [&](int32_t a, int32_t b) { return first[a] < first[b]; });
            TG_HIP_CHECK(hipMemcpyAsync(d_obn, old_by_new.data(), (size_t)ng * 4,
                                        hipMemcpyHostToDevice, s->stream));
        }

        DevPage outp;
        outp.n = ng;
        int grid = (ng + TG_BLOCK - 1) / TG_BLOCK;
        if (grid < 1) grid = 1;
        /* key channels */
        for (size_t c = 0; c < group_channels.size(); c++) {
            if (group_types[c] == TG_VARCHAR) {
                DevBlock b;
                tg_status st = emit_varchar_key(ng, (int)c, d_obn, grid, &b);
                if (st != TG_OK) return st;
                outp.blocks.push_back(b);
                continue;
            }
            DevBlock b;
            b.type = group_types[c];
            b.n = ng;
            TG_POOL_ALLOC(s, &b.data, (int64_t)(ng ? ng : 1) * b.elem_size());
            int64_t words = (ng + 63) / 64;
            TG_POOL_ALLOC(s, &b.valid, (words ? words : 1) * 8);
            TG_HIP_CHECK(hipMemsetAsync(b.valid, 0xFF, words * 8, s->stream));
            hipLaunchKernelGGL(k_emit_keys, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               t.keystore, t.n_words, d_obn, ng, (int)c,
                               (int32_t)b.type, b.data, b.valid);
            TG_HIP_CHECK(hipGetLastError());
            outp.blocks.push_back(b);
        }
        /* aggregate outputs */
        for (size_t a = 0; a < agg_state.size(); a++) {
            const KAgg& ag = agg_state[a];
            bool final_out = (step != TG_STEP_PARTIAL);
            if (ag.fn == TG_AGG_AVG_F64 && !final_out) {
                /* PARTIAL avg state: (count BIGINT, sum DOUBLE) channel pair */
                DevBlock bc; bc.type = TG_BIGINT; bc.n = ng;
                TG_POOL_ALLOC(s, &bc.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.cnt, d_obn, ng, (int64_t*)bc.data);
                outp.blocks.push_back(bc);
                DevBlock bs; bs.type = TG_DOUBLE; bs.n = ng;
                TG_POOL_ALLOC(s, &bs.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_f64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.sum, d_obn, ng, (double*)bs.data);
                outp.blocks.push_back(bs);
            }
            else if (ag.fn == TG_AGG_AVG_F64) {
                DevBlock b; b.type = TG_DOUBLE; b.n = ng;
                TG_POOL_ALLOC(s, &b.data, (int64_t)(ng ? ng : 1) * 8);
                int64_t words = (ng + 63) / 64;
                TG_POOL_ALLOC(s, &b.valid, (words ? words : 1) * 8);
                TG_HIP_CHECK(hipMemsetAsync(b.valid, 0xFF, words * 8, s->stream));
                hipLaunchKernelGGL(k_emit_avg, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.sum, ag.cnt, d_obn, ng, (double*)b.data, b.valid);
                outp.blocks.push_back(b);
            }
            else if (ag.fn == TG_AGG_SUM_F64_EXACT && final_out) {
                DevBlock b; b.type = TG_DOUBLE; b.n = ng;
                TG_POOL_ALLOC(s, &b.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_exact, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   (const unsigned long long*)ag.cnt,
                                   (const unsigned long long*)ag.sum, d_obn, ng,
                                   1.0 / ag.scale, (double*)b.data);
                outp.blocks.push_back(b);
            }
            else if (ag.fn == TG_AGG_SUM_F64_EXACT) {
                /* PARTIAL state: (lo BIGINT, hi BIGINT) channel pair */
                DevBlock bl; bl.type = TG_BIGINT; bl.n = ng;
                TG_POOL_ALLOC(s, &bl.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.cnt, d_obn, ng, (int64_t*)bl.data);
                outp.blocks.push_back(bl);
                DevBlock bh; bh.type = TG_BIGINT; bh.n = ng;
                TG_POOL_ALLOC(s, &bh.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   (const long long*)ag.sum, d_obn, ng, (int64_t*)bh.data);
                outp.blocks.push_back(bh);
            }
            else if (ag.fn == TG_AGG_MIN_I64 || ag.fn == TG_AGG_MAX_I64) {
                /* state is sign-bit-biased u64; emit the plain int64 value
                   (PARTIAL and FINAL states are identical) */
                DevBlock b; b.type = TG_BIGINT; b.n = ng;
                TG_POOL_ALLOC(s, &b.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_i64_biased, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.cnt, d_obn, ng, (int64_t*)b.data);
                outp.blocks.push_back(b);
            }
            else if (ag.fn == TG_AGG_SUM_F64) {
                DevBlock b; b.type = TG_DOUBLE; b.n = ng;
                TG_POOL_ALLOC(s, &b.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_f64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.sum, d_obn, ng, (double*)b.data);
                outp.blocks.push_back(b);
            }
            else {
                DevBlock b; b.type = TG_BIGINT; b.n = ng;
                TG_POOL_ALLOC(s, &b.data, (int64_t)(ng ? ng : 1) * 8);
                hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                                   ag.cnt, d_obn, ng, (int64_t*)b.data);
                outp.blocks.push_back(b);
            }
            TG_HIP_CHECK(hipGetLastError());
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_obn);
        if (pa && step == TG_STEP_PARTIAL) {
            /* flush accounting: unique groups vs rows in (onFlush analog) */
            pa->on_flush(pa_bytes, pa_rows, ng, 1);
            pa_bytes = pa_rows = 0;
        }
        stage_output(std::move(outp));
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        if (input_finished && !emitted && !mark_distinct) {
            emitted = true;
            tg_status st = emit();
            if (st != TG_OK) return st;
        }
        emit_staged(out, finished);
        return TG_OK;
    }

    ~HashAggOp() override
    {
        if (t.state) tg_pool_free(s, t.state);
        if (t.varstore) (void)hipFree(t.varstore);
        if (t.var_cursor) (void)hipFree(t.var_cursor);
        if (d_group_types) (void)hipFree(d_group_types);
        if (t.keystore) tg_pool_free(s, t.keystore);
        if (t.first_row) tg_pool_free(s, t.first_row);
        if (t.counter) tg_pool_free(s, t.counter);
        for (auto& a : agg_state) {
            if (a.sum) tg_pool_free(s, a.sum);
            if (a.cnt) tg_pool_free(s, a.cnt);
        }
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

extern "C" tg_status tg_hash_aggregation_create(tg_session* s,
    const int32_t* group_channels, int32_t n_group_channels,
    const int32_t* group_types, const tg_agg_spec* aggs, int32_t n_aggs,
    int32_t step, tg_operator** out)
{
    if (!s || !out || n_group_channels < 0 || n_group_channels > 7 ||
        n_aggs > MAX_AGGS || (n_group_channels > 0 && (!group_channels || !group_types))) {
        TG_SET_ERR("invalid hash aggregation spec (0..7 group channels, <=%d aggs)", MAX_AGGS);
        return TG_ERR_INVALID_ARG;
    }
    auto* op = new HashAggOp();
    op->s = s;
    if (n_group_channels > 0) {
        op->group_channels.assign(group_channels, group_channels + n_group_channels);
        for (int i = 0; i < n_group_channels; i++)
            op->group_types.push_back((tg_type)group_types[i]);
    }
    op->step = (tg_agg_step)step;
    for (int a = 0; a < n_aggs; a++) {
        op->aggs.push_back(aggs[a]);
        KAgg k{};
        k.fn = aggs[a].fn;
        k.in_ch = aggs[a].input_channel;
        k.mask_a = aggs[a].mask_gt_a;
        k.mask_b = aggs[a].mask_gt_b;
        bool needs_sum = (k.fn == TG_AGG_SUM_F64 || k.fn == TG_AGG_AVG_F64 ||
                          k.fn == TG_AGG_SUM_F64_EXACT);
        bool needs_cnt = (k.fn != TG_AGG_SUM_F64);
        k.scale = 1.0;
        for (int32_t sp_ = 0; sp_ < aggs[a].scale_pow; sp_++) k.scale *= 2.0;
        int64_t mg = 1 << 16;
        if (needs_sum) {
            if (tg_pool_alloc(s, (void**)&k.sum, mg * 8) != TG_OK) { delete op; return TG_ERR_OOM; }
        }
        if (needs_cnt) {
            if (tg_pool_alloc(s, (void**)&k.cnt, mg * 8) != TG_OK) { delete op; return TG_ERR_OOM; }
        }
        op->agg_state.push_back(k);
    }
    if (!op->group_types.empty()) {
        if (hipMalloc(&op->d_group_types, op->group_types.size() * 4) != hipSuccess ||
            hipMemcpy(op->d_group_types, group_types, op->group_types.size() * 4,
                      hipMemcpyHostToDevice) != hipSuccess) {
            delete op;
            TG_SET_ERR("alloc group types");
            return TG_ERR_OOM;
        }
    }
    bool any_var = false;
    for (tg_type gt : op->group_types) any_var |= (gt == TG_VARCHAR);
    if (any_var) {
        op->t.var_capacity = 1 << 20;
        if (hipMalloc(&op->t.varstore, op->t.var_capacity) != hipSuccess ||
            hipMalloc(&op->t.var_cursor, 8) != hipSuccess ||
            hipMemset(op->t.var_cursor, 0, 8) != hipSuccess) {
            delete op;
            TG_SET_ERR("alloc varchar key store");
            return TG_ERR_OOM;
        }
    }
    tg_status st = op->init_table(1 << 17, 1 << 16);
    if (st != TG_OK) { delete op; return st; }
    *out = op;
    return TG_OK;
}

/* ===== dense-range single-BIGINT-key aggregation =====================
 * When the group key is known to lie in a dense range (e.g. generated
 * custkeys 1..150k*SF), the hash table degenerates to direct array state:
 * one atomic per row, no probes, no keystore. A hardware-first variant of
 * BigintGroupByHash for the planner's "key statistics known" case; output
 * groups emit in KEY order (SQL-level parity: pipelines re-order anyway).
 * COUNT_STAR / COUNT_COL / SUM_I64 only (the use cases are count shapes). */

__global__ void k_dense_update(const KColH key, int64_t n, int64_t key_min,
                               int64_t range, long long* __restrict__ state,
                               const KColH val, int fn, double scale)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        if (kcol_is_null(key, i)) continue;
        int64_t k = (int64_t)kcol_word(key, i) - key_min;
        if (k < 0 || k >= range) continue;   /* guarded (caller promises) */
        if (fn == TG_AGG_SUM_F64_EXACT) {
            /* 128-bit fixed-point direct-array state (2 words per key):
             * same carry-propagating atomics as the hash/streaming paths */
            if (kcol_is_null(val, i)) continue;
            double y = ((const double*)val.data)[i] * scale;
            __int128 yi = (__int128)(long long)y;
            unsigned long long lov = (unsigned long long)(unsigned __int128)yi;
            unsigned long long hiv =
                (unsigned long long)((unsigned __int128)yi >> 64);
            unsigned long long old =
                atomicAdd((unsigned long long*)&state[2 * k], lov);
            unsigned long long carry = (old + lov) < lov ? 1ull : 0ull;
            atomicAdd((unsigned long long*)&state[2 * k + 1], hiv + carry);
            continue;
        }
        long long add = 1;
        if (fn == TG_AGG_SUM_I64 || fn == TG_AGG_COUNT_COL)
            add = (fn == TG_AGG_COUNT_COL)
                ? (kcol_is_null(val, i) ? 0 : 1)
                : (kcol_is_null(val, i) ? 0 : ((const long long*)val.data)[i]);
        if (add) atomicAdd((unsigned long long*)&state[k], (unsigned long long)add);
    }
}

/* LDS-privatized variant for small ranges (<= 8192 entries): per-block
 * shared histogram, one global atomic per present entry per block — fixes
 * hot-entry contention (Q13's ~64-value count histogram over 13.5M rows
 * measured 25.3 ms with direct global atomics). */
__global__ void k_dense_update_lds(const KColH key, int64_t n, int64_t key_min,
                                   int64_t range, long long* __restrict__ state,
                                   const KColH val, int fn)
{
    extern __shared__ long long h[];
    for (int64_t i = threadIdx.x; i < range; i += blockDim.x) h[i] = 0;
    __syncthreads();
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        if (kcol_is_null(key, i)) continue;
        int64_t k = (int64_t)kcol_word(key, i) - key_min;
        if (k < 0 || k >= range) continue;
        long long add = 1;
        if (fn == TG_AGG_SUM_I64 || fn == TG_AGG_COUNT_COL)
            add = (fn == TG_AGG_COUNT_COL)
                ? (kcol_is_null(val, i) ? 0 : 1)
                : (kcol_is_null(val, i) ? 0 : ((const long long*)val.data)[i]);
        if (add) atomicAdd((unsigned long long*)&h[k], (unsigned long long)add);
    }
    __syncthreads();
    for (int64_t k = threadIdx.x; k < range; k += blockDim.x)
        if (h[k]) atomicAdd((unsigned long long*)&state[k],
                            (unsigned long long)h[k]);
}

/* LDS-privatized 128-bit exact sum for small ranges: per-block (lo, hi)
 * pair per key with the carry-propagating LDS atomics, then ONE global
 * atomic pair per present key per block. Q9's 175 (nation, year) groups
 * over ~30M rows cost 17.5 ms through the generic hash aggregation's
 * hot-group atomics (profiles/profq9). */
__global__ void k_dense_update_lds_exact(const KColH key, int64_t n,
                                         int64_t key_min, int64_t range,
                                         long long* __restrict__ state,
                                         const KColH val, double scale)
{
    extern __shared__ long long h[];   /* [2*range]: lo, hi interleaved */
    for (int64_t i = threadIdx.x; i < 2 * range; i += blockDim.x) h[i] = 0;
    __syncthreads();
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        if (kcol_is_null(key, i) || kcol_is_null(val, i)) continue;
        int64_t k = (int64_t)kcol_word(key, i) - key_min;
        if (k < 0 || k >= range) continue;
        double y = ((const double*)val.data)[i] * scale;
        __int128 yi = (__int128)(long long)y;
        unsigned long long lov = (unsigned long long)(unsigned __int128)yi;
        unsigned long long hiv = (unsigned long long)((unsigned __int128)yi >> 64);
        unsigned long long old =
            atomicAdd((unsigned long long*)&h[2 * k], lov);
        unsigned long long carry = (old + lov) < lov ? 1ull : 0ull;
        atomicAdd((unsigned long long*)&h[2 * k + 1], hiv + carry);
    }
    __syncthreads();
    for (int64_t k = threadIdx.x; k < range; k += blockDim.x) {
        unsigned long long lov = (unsigned long long)h[2 * k];
        unsigned long long hiv = (unsigned long long)h[2 * k + 1];
        if (!lov && !hiv) continue;
        unsigned long long old =
            atomicAdd((unsigned long long*)&state[2 * k], lov);
        unsigned long long carry = (old + lov) < lov ? 1ull : 0ull;
        atomicAdd((unsigned long long*)&state[2 * k + 1], hiv + carry);
    }
}

__global__ void k_dense_present(const long long* __restrict__ state, int64_t range,
                                int32_t* __restrict__ chunk_counts, int64_t nchunks,
                                int64_t chunk, int words)
{
    int64_t c = (int64_t)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    if (c >= nchunks) return;
    int lane = threadIdx.x % 64;
    int64_t lo = c * chunk, hi = min(lo + chunk, range);
    int32_t cnt = 0;
    for (int64_t i = lo + lane; i < hi; i += 64)
        cnt += (state[i * words] != 0 ||
                (words == 2 && state[i * words + 1] != 0));
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1) cnt += __shfl_xor(cnt, off, 64);
    if (lane == 0) chunk_counts[c] = cnt;
}

__global__ void k_dense_emit(const long long* __restrict__ state, int64_t range,
                             int64_t key_min, const int32_t* __restrict__ offs,
                             int64_t nchunks, int64_t chunk,
                             int64_t* __restrict__ out_keys,
                             int64_t* __restrict__ out_vals)
{
    int64_t c = blockIdx.x;
    if (c >= nchunks) return;
    if (threadIdx.x >= 64) return;
    int lane = threadIdx.x;
    int64_t lo = c * chunk, hi = min(lo + chunk, range);
    int32_t run = offs[c];
    for (int64_t g = lo; g < hi; g += 64) {
        int64_t i = g + lane;
        bool p = (i < hi) && state[i] != 0;
        unsigned long long b = __ballot(p);
        int before = __popcll(b & ((1ull << lane) - 1ull));
        if (p) {
            out_keys[run + before] = key_min + i;
            out_vals[run + before] = state[i];
        }
        run += __popcll(b);
    }
}

/* exact-sum variant: 2-word state, emits the correctly-rounded double */
__global__ void k_dense_emit_exact(const long long* __restrict__ state,
                                   int64_t range, int64_t key_min,
                                   const int32_t* __restrict__ offs,
                                   int64_t nchunks, int64_t chunk,
                                   double inv_scale,
                                   int64_t* __restrict__ out_keys,
                                   double* __restrict__ out_vals)
{
    int64_t c = blockIdx.x;
    if (c >= nchunks) return;
    if (threadIdx.x >= 64) return;
    int lane = threadIdx.x;
    int64_t lo = c * chunk, hi = min(lo + chunk, range);
    int32_t run = offs[c];
    for (int64_t g = lo; g < hi; g += 64) {
        int64_t i = g + lane;
        bool p = (i < hi) && (state[2 * i] != 0 || state[2 * i + 1] != 0);
        unsigned long long b = __ballot(p);
        int before = __popcll(b & ((1ull << lane) - 1ull));
        if (p) {
            unsigned __int128 v =
                ((unsigned __int128)(unsigned long long)state[2 * i + 1] << 64) |
                (unsigned long long)state[2 * i];
            __int128 sv = (__int128)v;
            out_keys[run + before] = key_min + i;
            out_vals[run + before] = (double)sv * inv_scale;
        }
        run += __popcll(b);
    }
}

struct DenseAggOp : tg_operator {
    int32_t key_channel = 0;
    int64_t key_min = 0, range = 0;
    tg_agg_spec agg{};
    long long* state = nullptr;
    double scale = 1.0;     /* SUM_F64_EXACT fixed-point scale (2^scale_pow) */
    int words = 1;          /* state words per key (2 for SUM_F64_EXACT) */
    bool emitted = false;

    tg_status add_input(const tg_page* page) override
    {
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        KColH* d_all = nullptr;
        st = make_kcols(s, in, nullptr, (int)in.blocks.size(), &d_all);
        if (st != TG_OK) { tg_free_page(s, &in); return st; }
        std::vector<KColH> h(in.blocks.size());
        TG_HIP_CHECK(hipMemcpy(h.data(), d_all, h.size() * sizeof(KColH),
                               hipMemcpyDeviceToHost));
        KColH kc = h[key_channel];
        KColH vc = (agg.input_channel >= 0) ? h[agg.input_channel] : kc;
        if (range <= 4096 && words == 2) {
            hipLaunchKernelGGL(k_dense_update_lds_exact,
                               dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                               range * 16, s->stream, kc, in.n, key_min,
                               range, state, vc, scale);
        }
        else if (range <= 8192 && words == 1) {
            hipLaunchKernelGGL(k_dense_update_lds, dim3(tg_grid_for(in.n)),
                               dim3(TG_BLOCK), range * 8, s->stream, kc, in.n,
                               key_min, range, state, vc, agg.fn);
        }
        else {
            hipLaunchKernelGGL(k_dense_update, dim3(tg_grid_for(in.n)),
                               dim3(TG_BLOCK), 0, s->stream, kc, in.n,
                               key_min, range, state, vc, agg.fn, scale);
        }
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_all);
        tg_free_page(s, &in);
        return TG_OK;
    }

    tg_status emit()
    {
        constexpr int64_t CH = 16384;
        int64_t nchunks = (range + CH - 1) / CH;
        int32_t* d_offs = nullptr;
        int32_t* d_total = nullptr;
        TG_POOL_ALLOC(s, &d_offs, (nchunks ? nchunks : 1) * 4);
        TG_POOL_ALLOC(s, &d_total, 4);
        int wpb = TG_BLOCK / 64;
        hipLaunchKernelGGL(k_dense_present,
                           dim3((uint32_t)((nchunks + wpb - 1) / wpb)),
                           dim3(TG_BLOCK), 0, s->stream, state, range, d_offs,
                           nchunks, CH, words);
        TG_HIP_CHECK(hipGetLastError());
        tg_status st = run_scan_i32(s, d_offs, nchunks, d_total);
        if (st != TG_OK) return st;
        int32_t total = 0;
        TG_HIP_CHECK(hipMemcpy(&total, d_total, 4, hipMemcpyDeviceToHost));
        DevPage outp;
        outp.n = total;
        outp.blocks.resize(2);
        DevBlock& bk = outp.blocks[0];
        bk.type = TG_BIGINT; bk.n = total;
        TG_POOL_ALLOC(s, &bk.data, (int64_t)(total ? total : 1) * 8);
        DevBlock& bv = outp.blocks[1];
        bv.type = (agg.fn == TG_AGG_SUM_F64_EXACT) ? TG_DOUBLE : TG_BIGINT;
        bv.n = total;
        TG_POOL_ALLOC(s, &bv.data, (int64_t)(total ? total : 1) * 8);
        if (agg.fn == TG_AGG_SUM_F64_EXACT) {
            hipLaunchKernelGGL(k_dense_emit_exact, dim3((uint32_t)nchunks),
                               dim3(64), 0, s->stream, state, range, key_min,
                               d_offs, nchunks, CH, 1.0 / scale,
                               (int64_t*)bk.data, (double*)bv.data);
        }
        else {
            hipLaunchKernelGGL(k_dense_emit, dim3((uint32_t)nchunks), dim3(64),
                               0, s->stream, state, range, key_min, d_offs,
                               nchunks, CH, (int64_t*)bk.data,
                               (int64_t*)bv.data);
        }
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_offs);
        tg_pool_free(s, d_total);
        stage_output(std::move(outp));
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        if (input_finished && !emitted) {
            emitted = true;
            tg_status st = emit();
            if (st != TG_OK) return st;
        }
        emit_staged(out, finished);
        return TG_OK;
    }

    ~DenseAggOp() override
    {
        if (state) tg_pool_free(s, state);
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

extern "C" tg_status tg_dense_aggregation_create(tg_session* s,
    int32_t key_channel, int64_t key_min, int64_t key_max,
    const tg_agg_spec* agg, tg_operator** out)
{
    if (!s || !out || !agg || key_max < key_min ||
        key_max - key_min + 1 > (1ll << 34)) {
        TG_SET_ERR("invalid dense aggregation spec (range <= 2^34)");
        return TG_ERR_INVALID_ARG;
    }
    if (agg->fn != TG_AGG_COUNT_STAR && agg->fn != TG_AGG_COUNT_COL &&
        agg->fn != TG_AGG_SUM_I64 && agg->fn != TG_AGG_SUM_F64_EXACT) {
        TG_SET_ERR("dense aggregation supports COUNT/SUM_I64/SUM_F64_EXACT");
        return TG_ERR_UNSUPPORTED;
    }
    auto* op = new DenseAggOp();
    op->s = s;
    op->key_channel = key_channel;
    op->key_min = key_min;
    op->range = key_max - key_min + 1;
    op->agg = *agg;
    if (agg->fn == TG_AGG_SUM_F64_EXACT) {
        op->words = 2;
        for (int32_t sp_ = 0; sp_ < agg->scale_pow; sp_++) op->scale *= 2.0;
    }
    if (tg_pool_alloc(s, (void**)&op->state, op->range * 8 * op->words) != TG_OK) {
        delete op;
        return TG_ERR_OOM;
    }
    (void)hipMemsetAsync(op->state, 0, op->range * 8 * op->words, s->stream);
    (void)hipStreamSynchronize(s->stream);
    *out = op;
    return TG_OK;
}

/* MarkDistinctOperator analog (operator/MarkDistinctOperator.java): appends
 * a BOOLEAN channel marking each row's group-first occurrence over the
 * given key channels. Streaming:each addInput emits its page + flags. */
extern "C" tg_status tg_mark_distinct_create(tg_session* s,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* key_types, tg_operator** out)
{
    tg_status st = tg_hash_aggregation_create(s, key_channels, n_key_channels,
                                              key_types, nullptr, 0,
                                              TG_STEP_SINGLE, out);
    if (st != TG_OK) return st;
    static_cast<HashAggOp*>(*out)->mark_distinct = true;
    return TG_OK;
}

/* ---- adaptive partial aggregation ABI ---- */
extern "C" tg_status tg_pa_controller_create(int64_t max_partial_bytes,
    double unique_rows_ratio_threshold, tg_pa_controller** out)
{
    if (!out || max_partial_bytes <= 0 || unique_rows_ratio_threshold <= 0.0) {
        TG_SET_ERR("invalid partial aggregation controller spec");
        return TG_ERR_INVALID_ARG;
    }
    auto* c = new tg_pa_controller();
    c->max_partial_bytes = max_partial_bytes;
    c->unique_ratio_threshold = unique_rows_ratio_threshold;
    *out = c;
    return TG_OK;
}

extern "C" void tg_pa_controller_close(tg_pa_controller* c) { delete c; }

extern "C" int32_t tg_pa_controller_disabled(tg_pa_controller* c)
{
    return (c && c->disabled.load(std::memory_order_relaxed)) ? 1 : 0;
}

extern "C" tg_status tg_pa_controller_on_flush(tg_pa_controller* c,
    int64_t bytes, int64_t rows, int64_t unique_rows, int32_t have_unique)
{
    if (!c) {
        TG_SET_ERR("null controller");
        return TG_ERR_INVALID_ARG;
    }
    c->on_flush(bytes, rows, unique_rows, have_unique);
    return TG_OK;
}

extern "C" tg_status tg_hash_aggregation_set_controller(tg_operator* op,
    tg_pa_controller* c)
{
    auto* h = dynamic_cast<HashAggOp*>(op);
    if (!h || !c) {
        TG_SET_ERR("controller requires a hash aggregation operator");
        return TG_ERR_INVALID_ARG;
    }
    if (h->step != TG_STEP_PARTIAL || h->mark_distinct) {
        TG_SET_ERR("adaptive partial aggregation applies to the PARTIAL step");
        return TG_ERR_INVALID_ARG;
    }
    for (tg_type ty : h->group_types) {
        if (ty == TG_VARCHAR) {
            TG_SET_ERR("adaptive partial aggregation: VARCHAR group keys "
                       "unsupported (pass-through copies fixed-width keys)");
            return TG_ERR_UNSUPPORTED;
        }
    }
    for (const tg_agg_spec& a : h->aggs) {
        if (a.mask_gt_a >= 0 && a.mask_gt_a != a.mask_gt_b) {
            TG_SET_ERR("adaptive partial aggregation: masked aggregates "
                       "unsupported");
            return TG_ERR_UNSUPPORTED;
        }
    }
    h->pa = c;
    return TG_OK;
}
