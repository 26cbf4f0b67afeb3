/* q1.hip — fused TPC-H Q1 scan+filter+aggregate kernel for gfx950.
 *
 * Replaces, in one HBM pass, the reference pipeline
 *   ScanFilterAndProjectOperator (ColumnarFilter shipdate<=cutoff,
 *   sql/gen/columnar/ColumnarFilter.java:42-52) ->
 *   projections extprice*(1-disc), *(1+tax) (project/PageProcessor.java:274-305) ->
 *   HashAggregationOperator over (returnflag, linestatus)
 *   (InMemoryHashAggregationBuilder.java:140-158, DoubleSumAggregation.java:37-45,
 *    DoubleAverageAggregations.java:38-63, CountAggregation).
 *
 * Algorithmic traffic: 38 B/row (shipdate i32 + 4 f64 + 2 u8); HBM-bound,
 * no MFMA (no dense contraction here). Roofline ceiling 8 TB/s / 38 B =
 * 210 Grow/s per GPU.
 *
 * FP sums are error-free fixed-point: every addend x has
 * ulp >= 2^-43 (money columns, x >= 2^9) or >= 2^-59 (discount), so
 * y = x * 2^scale is an exact u64 integer; lanes accumulate y into u128.
 * The final sums are the correctly-rounded exact sums, order-independent —
 * bit-equal to oracle o_q1_exact. sum(quantity) and count are exact integers.
 *
 * Parity-mode kernel (k_q1_naive_seq): single thread, reference sequential
 * order — bit-equal to oracle o_q1_naive for page-sized operator tests.
 */
#include "common.h"
#include <cstdlib>

#define NCOMBO 6
#define S43 8796093022208.0        /* 2^43 */
#define S59 576460752303423488.0   /* 2^59 */

/* per-block partial layout: per combo 10 u64 words:
 * base(2) dp(2) ch(2) disc(2) qty(1) cnt(1) => 60 words per block */
#define PARTIAL_WORDS (NCOMBO * 10)

struct lane_acc {
    u128 base, dp, ch, disc;
    unsigned long long qty;
    unsigned long long cnt;
};

__device__ static inline void wave_reduce_u128(u128& v)
{
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1) {
        unsigned long long lo = __shfl_xor(v.lo, off, 64);
        unsigned long long hi = __shfl_xor(v.hi, off, 64);
        unsigned long long nl = v.lo + lo;
        v.hi = v.hi + hi + (nl < lo);
        v.lo = nl;
    }
}

__device__ static inline void wave_reduce_u64(unsigned long long& v)
{
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1)
        v += __shfl_xor(v, off, 64);
}

template <int R>   /* rows per lane per iteration (2 or 4) */
__global__ __launch_bounds__(TG_BLOCK, R == 2 ? 3 : 2)   /* 2nd arg: waves/SIMD */
void k_q1_fused(int64_t n, const int32_t* __restrict__ shipdate,
                const double* __restrict__ qty, const double* __restrict__ extprice,
                const double* __restrict__ disc, const double* __restrict__ tax,
                const uint8_t* __restrict__ rflag, const uint8_t* __restrict__ lstatus,
                int32_t cutoff, unsigned long long* __restrict__ partials)
{
    lane_acc acc[NCOMBO];
    #pragma unroll
    for (int c = 0; c < NCOMBO; c++) { acc[c].qty = 0; acc[c].cnt = 0; }

    const int64_t stride = (int64_t)gridDim.x * blockDim.x;   /* lanes total */
    const int64_t gid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t ngrp = n / R;

    for (int64_t i = gid; i < ngrp; i += stride) {
        const int64_t r = R * i;
        double vq[R], ve[R], vd[R], vt[R];
        int32_t sd[R];
        uint8_t rf[R], ls[R];
        #pragma unroll
        for (int j = 0; j < R; j += 2) {   /* 16 B/lane vector loads */
            *reinterpret_cast<double2*>(vq + j) = *reinterpret_cast<const double2*>(qty + r + j);
            *reinterpret_cast<double2*>(ve + j) = *reinterpret_cast<const double2*>(extprice + r + j);
            *reinterpret_cast<double2*>(vd + j) = *reinterpret_cast<const double2*>(disc + r + j);
            *reinterpret_cast<double2*>(vt + j) = *reinterpret_cast<const double2*>(tax + r + j);
        }
        if (R == 4) {
            *reinterpret_cast<int4*>(sd) = *reinterpret_cast<const int4*>(shipdate + r);
            *reinterpret_cast<uchar4*>(rf) = *reinterpret_cast<const uchar4*>(rflag + r);
            *reinterpret_cast<uchar4*>(ls) = *reinterpret_cast<const uchar4*>(lstatus + r);
        }
        else {
            *reinterpret_cast<int2*>(sd) = *reinterpret_cast<const int2*>(shipdate + r);
            *reinterpret_cast<uchar2*>(rf) = *reinterpret_cast<const uchar2*>(rflag + r);
            *reinterpret_cast<uchar2*>(ls) = *reinterpret_cast<const uchar2*>(lstatus + r);
        }
        bool sel[R];
        int cb[R];
        unsigned long long yb[R], yp[R], yc[R], yd[R], yq[R];
        #pragma unroll
        for (int j = 0; j < R; j++) {
            sel[j] = sd[j] <= cutoff;
            cb[j] = rf[j] * 2 + ls[j];
            double dp = ve[j] * (1.0 - vd[j]);
            double ch = dp * (1.0 + vt[j]);
            yb[j] = (unsigned long long)(ve[j] * S43);
            yp[j] = (unsigned long long)(dp * S43);
            yc[j] = (unsigned long long)(ch * S43);
            yd[j] = (unsigned long long)(vd[j] * S59);
            yq[j] = (unsigned long long)vq[j];
        }
        #pragma unroll
        for (int c = 0; c < NCOMBO; c++) {
            bool any = false;
            #pragma unroll
            for (int j = 0; j < R; j++) any |= (sel[j] & (cb[j] == c));
            if (__any(any)) {
                /* combine R masked rows into ONE u64 per sum (< 2^63), then a
                 * single u128 add per sum per combo */
                unsigned long long tb = 0, tp = 0, tc = 0, td = 0, tq = 0;
                int tn = 0;
                #pragma unroll
                for (int j = 0; j < R; j++) {
                    unsigned long long k = (sel[j] & (cb[j] == c)) ? ~0ull : 0ull;
                    tb += yb[j] & k; tp += yp[j] & k; tc += yc[j] & k;
                    td += yd[j] & k; tq += yq[j] & k;
                    tn += (int)(k & 1);
                }
                acc[c].base.add(tb);
                acc[c].dp.add(tp);
                acc[c].ch.add(tc);
                acc[c].disc.add(td);
                acc[c].qty += tq;
                acc[c].cnt += tn;
            }
        }
    }
    /* leftover rows [ngrp*R, n) handled by the first lane. NOTE: combo index
     * must be compile-time constant per unrolled iteration — a runtime acc[c]
     * index demotes the whole accumulator array to scratch (measured:
     * 496 B/lane, 18% roofline). */
    for (int64_t r = ngrp * R; gid == 0 && r < n; r++) {
        if (shipdate[r] <= cutoff) {
            int ct = rflag[r] * 2 + lstatus[r];
            double dpv = extprice[r] * (1.0 - disc[r]);
            double chv = dpv * (1.0 + tax[r]);
            unsigned long long tb = (unsigned long long)(extprice[r] * S43);
            unsigned long long tp = (unsigned long long)(dpv * S43);
            unsigned long long tc = (unsigned long long)(chv * S43);
            unsigned long long td = (unsigned long long)(disc[r] * S59);
            unsigned long long tq = (unsigned long long)qty[r];
            #pragma unroll
            for (int c = 0; c < NCOMBO; c++) {
                if (ct == c) {
                    acc[c].base.add(tb);
                    acc[c].dp.add(tp);
                    acc[c].ch.add(tc);
                    acc[c].disc.add(td);
                    acc[c].qty += tq;
                    acc[c].cnt += 1;
                }
            }
        }
    }

    /* wave -> block -> global reduction (integers: order-free, deterministic) */
    __shared__ unsigned long long lds[TG_BLOCK / 64][PARTIAL_WORDS];
    const int wave = threadIdx.x / 64;
    const int lane = threadIdx.x % 64;
    #pragma unroll
    for (int c = 0; c < NCOMBO; c++) {
        wave_reduce_u128(acc[c].base);
        wave_reduce_u128(acc[c].dp);
        wave_reduce_u128(acc[c].ch);
        wave_reduce_u128(acc[c].disc);
        wave_reduce_u64(acc[c].qty);
        wave_reduce_u64(acc[c].cnt);
        if (lane == 0) {
            unsigned long long* w = lds[wave] + c * 10;
            w[0] = acc[c].base.lo; w[1] = acc[c].base.hi;
            w[2] = acc[c].dp.lo;   w[3] = acc[c].dp.hi;
            w[4] = acc[c].ch.lo;   w[5] = acc[c].ch.hi;
            w[6] = acc[c].disc.lo; w[7] = acc[c].disc.hi;
            w[8] = acc[c].qty;     w[9] = acc[c].cnt;
        }
    }
    __syncthreads();
    if (threadIdx.x < PARTIAL_WORDS) {
        const int f = threadIdx.x;          /* one word per thread */
        const int is_hi = ((f % 10) < 8) && (f % 2 == 1);
        unsigned long long lo = 0, carryhi = 0;
        if (!is_hi && (f % 10) < 8) {
            /* lo word of a u128 field: sum los with carry into hi below */
            unsigned long long hs = 0;
            #pragma unroll
            for (int w = 0; w < TG_BLOCK / 64; w++) {
                unsigned long long x = lds[w][f];
                lo += x;
                hs += (lo < x);
                carryhi = hs;
            }
            unsigned long long hisum = carryhi;
            #pragma unroll
            for (int w = 0; w < TG_BLOCK / 64; w++) hisum += lds[w][f + 1];
            partials[(int64_t)blockIdx.x * PARTIAL_WORDS + f] = lo;
            partials[(int64_t)blockIdx.x * PARTIAL_WORDS + f + 1] = hisum;
        }
        else if ((f % 10) >= 8) {
            unsigned long long s = 0;
            #pragma unroll
            for (int w = 0; w < TG_BLOCK / 64; w++) s += lds[w][f];
            partials[(int64_t)blockIdx.x * PARTIAL_WORDS + f] = s;
        }
        /* hi words written by their lo-word thread */
    }
}

/* final cross-block reduce: 36 (combo,field) owners × 7 block-chunks */
#define RED_CHUNKS 7
__global__ void k_q1_reduce(const unsigned long long* __restrict__ partials,
                            int nblocks, unsigned long long* __restrict__ out)
{
    __shared__ unsigned long long plo[NCOMBO * 6][RED_CHUNKS];
    __shared__ unsigned long long phi[NCOMBO * 6][RED_CHUNKS];
    int t = threadIdx.x;
    int fld = t % (NCOMBO * 6);
    int chunk = t / (NCOMBO * 6);
    if (chunk < RED_CHUNKS) {
        int c = fld / 6, f = fld % 6;
        int per = (nblocks + RED_CHUNKS - 1) / RED_CHUNKS;
        int b0 = chunk * per, b1 = min(b0 + per, nblocks);
        u128 s;
        if (f < 4) {
            for (int b = b0; b < b1; b++) {
                const unsigned long long* p = partials + (int64_t)b * PARTIAL_WORDS + c * 10 + f * 2;
                u128 v; v.lo = p[0]; v.hi = p[1];
                s.add128(v);
            }
        }
        else {
            for (int b = b0; b < b1; b++)
                s.lo += partials[(int64_t)b * PARTIAL_WORDS + c * 10 + 8 + (f - 4)];
        }
        plo[fld][chunk] = s.lo;
        phi[fld][chunk] = s.hi;
    }
    __syncthreads();
    if (t < NCOMBO * 6) {
        int c = t / 6, f = t % 6;
        u128 s;
        for (int k = 0; k < RED_CHUNKS; k++) {
            u128 v; v.lo = plo[t][k]; v.hi = phi[t][k];
            s.add128(v);
        }
        if (f < 4) {
            out[c * 10 + f * 2] = s.lo;
            out[c * 10 + f * 2 + 1] = s.hi;
        }
        else {
            out[c * 10 + 8 + (f - 4)] = s.lo;
        }
    }
}

/* parity-mode: the reference's single-driver sequential accumulation order
 * (bit-equal to oracle o_q1_naive). One thread; page-sized inputs only. */
__global__ void k_q1_naive_seq(int64_t n, const int32_t* shipdate, const double* qty,
                               const double* extprice, const double* disc,
                               const double* tax, const uint8_t* rflag,
                               const uint8_t* lstatus, int32_t cutoff,
                               double* sums /* [6][5]: qty,base,dp,ch,disc */,
                               long long* cnts /* [6] */)
{
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    for (int c = 0; c < NCOMBO; c++) {
        for (int f = 0; f < 5; f++) sums[c * 5 + f] = 0.0;
        cnts[c] = 0;
    }
    for (int64_t i = 0; i < n; i++) {
        if (shipdate[i] > cutoff) continue;
        int c = rflag[i] * 2 + lstatus[i];
        double dp = extprice[i] * (1.0 - disc[i]);
        double ch = dp * (1.0 + tax[i]);
        sums[c * 5 + 0] += qty[i];
        sums[c * 5 + 1] += extprice[i];
        sums[c * 5 + 2] += dp;
        sums[c * 5 + 3] += ch;
        sums[c * 5 + 4] += disc[i];
        cnts[c]++;
    }
}

struct q1_scratch {
    unsigned long long* partials = nullptr;
    unsigned long long* out = nullptr;
    double* naive_sums = nullptr;
    long long* naive_cnts = nullptr;
};
static q1_scratch g_scratch;   /* per-process; one session per process */

static tg_status ensure_scratch()
{
    if (!g_scratch.partials) {
        TG_HIP_CHECK(hipMalloc(&g_scratch.partials,
                               (size_t)TG_MAX_BLOCKS * PARTIAL_WORDS * 8));
        TG_HIP_CHECK(hipMalloc(&g_scratch.out, PARTIAL_WORDS * 8));
        TG_HIP_CHECK(hipMalloc(&g_scratch.naive_sums, NCOMBO * 5 * 8));
        TG_HIP_CHECK(hipMalloc(&g_scratch.naive_cnts, NCOMBO * 8));
    }
    return TG_OK;
}

extern "C" tg_status tg_q1_run(tg_session* s, const tg_tpch_lineitem_cols* cols,
                               int32_t cutoff, tg_q1_result* out)
{
    if (!s || !cols || !out) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    tg_status st = ensure_scratch();
    if (st != TG_OK) return st;
    int64_t n = cols->row_count;
    /* variant/grid sweep hooks (defaults measured on MI355X: R=4, 1536
     * blocks -> 6.0 TB/s = 75% of the 8 TB/s roofline; profiles/r01) */
    static int R = [] { const char* e = getenv("TG_Q1_R"); return e ? atoi(e) : 4; }();
    static int blocks_env = [] { const char* e = getenv("TG_Q1_BLOCKS"); return e ? atoi(e) : 0; }();
    int grid = tg_grid_for(n, R);
    if (grid > 1536) grid = 1536;
    if (blocks_env > 0) grid = blocks_env;
    if (grid > TG_MAX_BLOCKS) grid = TG_MAX_BLOCKS;   /* partials buffer bound */
    TG_HIP_CHECK(hipEventRecord(s->ev_start, s->stream));
    if (R == 4)
        hipLaunchKernelGGL(k_q1_fused<4>, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                           n, cols->shipdate, cols->quantity, cols->extendedprice,
                           cols->discount, cols->tax, cols->returnflag, cols->linestatus,
                           cutoff, g_scratch.partials);
    else
        hipLaunchKernelGGL(k_q1_fused<2>, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                           n, cols->shipdate, cols->quantity, cols->extendedprice,
                           cols->discount, cols->tax, cols->returnflag, cols->linestatus,
                           cutoff, g_scratch.partials);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipEventRecord(s->ev_stop, s->stream));
    hipLaunchKernelGGL(k_q1_reduce, dim3(1), dim3(256), 0, s->stream,
                       g_scratch.partials, grid, g_scratch.out);
    TG_HIP_CHECK(hipGetLastError());
    unsigned long long h[PARTIAL_WORDS];
    TG_HIP_CHECK(hipMemcpyAsync(h, g_scratch.out, sizeof(h), hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    float ms = 0;
    TG_HIP_CHECK(hipEventElapsedTime(&ms, s->ev_start, s->ev_stop));

    memset(out, 0, sizeof(*out));
    memcpy(out->raw, h, sizeof(h));
    for (int c = 0; c < NCOMBO; c++) {
        const unsigned long long* w = h + c * 10;
        __int128 base = ((__int128)w[1] << 64) | w[0];
        __int128 dp   = ((__int128)w[3] << 64) | w[2];
        __int128 ch   = ((__int128)w[5] << 64) | w[4];
        __int128 dc   = ((__int128)w[7] << 64) | w[6];
        out->sum_base[c] = (double)base / S43;
        out->sum_disc_price[c] = (double)dp / S43;
        out->sum_charge[c] = (double)ch / S43;
        out->sum_disc[c] = (double)dc / S59;
        out->sum_qty[c] = (double)(long long)w[8];
        out->count[c] = (int64_t)w[9];
        if (out->count[c] > 0) {
            out->avg_qty[c] = out->sum_qty[c] / (double)out->count[c];
            out->avg_price[c] = out->sum_base[c] / (double)out->count[c];
            out->avg_disc[c] = out->sum_disc[c] / (double)out->count[c];
        }
    }
    out->elapsed_ms = (double)ms;
    return TG_OK;
}

/* parity-mode entry (naive sequential; page-sized inputs) */
extern "C" tg_status tg_q1_run_naive(tg_session* s, const tg_tpch_lineitem_cols* cols,
                                     int32_t cutoff, tg_q1_result* out)
{
    if (!s || !cols || !out) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    tg_status st = ensure_scratch();
    if (st != TG_OK) return st;
    hipLaunchKernelGGL(k_q1_naive_seq, dim3(1), dim3(64), 0, s->stream,
                       cols->row_count, cols->shipdate, cols->quantity,
                       cols->extendedprice, cols->discount, cols->tax,
                       cols->returnflag, cols->linestatus, cutoff,
                       g_scratch.naive_sums, g_scratch.naive_cnts);
    TG_HIP_CHECK(hipGetLastError());
    double hs[NCOMBO * 5];
    long long hc[NCOMBO];
    TG_HIP_CHECK(hipMemcpyAsync(hs, g_scratch.naive_sums, sizeof(hs), hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipMemcpyAsync(hc, g_scratch.naive_cnts, sizeof(hc), hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    memset(out, 0, sizeof(*out));
    for (int c = 0; c < NCOMBO; c++) {
        out->sum_qty[c] = hs[c * 5 + 0];
        out->sum_base[c] = hs[c * 5 + 1];
        out->sum_disc_price[c] = hs[c * 5 + 2];
        out->sum_charge[c] = hs[c * 5 + 3];
        out->sum_disc[c] = hs[c * 5 + 4];
        out->count[c] = hc[c];
        if (hc[c] > 0) {
            out->avg_qty[c] = out->sum_qty[c] / (double)hc[c];
            out->avg_price[c] = out->sum_base[c] / (double)hc[c];
            out->avg_disc[c] = out->sum_disc[c] / (double)hc[c];
        }
    }
    return TG_OK;
}
