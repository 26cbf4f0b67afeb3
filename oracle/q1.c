/* q1.c — CPU oracle for TPC-H Q1 over the lineitem hot-path columns, plus the
 * timed CPU-baseline leg. ORACLE / TEST INFRASTRUCTURE ONLY.
 *
 * Restates the reference pipeline for Q1:
 *   filter (ColumnarFilter: shipdate <= cutoff, selection vector) ->
 *   project (extprice*(1-disc), extprice*(1-disc)*(1+tax); left-assoc IEEE) ->
 *   FlatGroupByHash group ids in row order over (returnflag, linestatus) ->
 *   accumulators DoubleSumAggregation.java:37-45 / DoubleAverageAggregations
 *   .java:38-63 / CountAggregation, sequential per-driver order.
 * Two legs:
 *   naive  — single sequential pass, the reference's exact accumulation order
 *            for a single-driver execution (bit-parity target for the GPU
 *            parity-mode kernel on page-sized inputs);
 *   exact  — correctly-rounded exact sums via 2^43 / 2^59 fixed-point
 *            (bit-parity target for the full-size fused GPU kernel).
 * Groups are keyed here by combo = returnflag*2 + linestatus (6 slots) and
 * reported with first-occurrence group ids so tests can check both shapes.
 */
#include <stdint.h>
#include <string.h>
#include <stdlib.h>
#ifdef _OPENMP
#include <omp.h>
#endif

#define EXPORT __attribute__((visibility("default")))

typedef struct {
    /* indexed by combo = rf*2 + ls (rf: 0=A 1=N 2=R; ls: 0=F 1=O) */
    double sum_qty[6], sum_base[6], sum_disc_price[6], sum_charge[6], sum_disc[6];
    double avg_qty[6], avg_price[6], avg_disc[6];
    int64_t count[6];
    int32_t group_id_by_combo[6];   /* row-order first-occurrence ids, -1 if absent */
    int32_t n_groups;
} o_q1_result;

static void q1_finish(o_q1_result* r)
{
    for (int c = 0; c < 6; c++) {
        if (r->count[c] > 0) {
            r->avg_qty[c]   = r->sum_qty[c]  / (double)r->count[c];
            r->avg_price[c] = r->sum_base[c] / (double)r->count[c];
            r->avg_disc[c]  = r->sum_disc[c] / (double)r->count[c];
        }
    }
}

/* naive leg: single-driver sequential order */
EXPORT void o_q1_naive(int64_t n, const int32_t* shipdate, const double* qty,
                       const double* extprice, const double* discount, const double* tax,
                       const uint8_t* rflag, const uint8_t* lstatus,
                       int32_t cutoff, o_q1_result* r)
{
    memset(r, 0, sizeof(*r));
    for (int c = 0; c < 6; c++) r->group_id_by_combo[c] = -1;
    int32_t next_gid = 0;
    for (int64_t i = 0; i < n; i++) {
        if (shipdate[i] > cutoff) continue;
        int c = rflag[i] * 2 + lstatus[i];
        if (r->group_id_by_combo[c] < 0) r->group_id_by_combo[c] = next_gid++;
        double dp = extprice[i] * (1.0 - discount[i]);
        double ch = dp * (1.0 + tax[i]);
        r->sum_qty[c] += qty[i];
        r->sum_base[c] += extprice[i];
        r->sum_disc_price[c] += dp;
        r->sum_charge[c] += ch;
        r->sum_disc[c] += discount[i];
        r->count[c]++;
    }
    r->n_groups = next_gid;
    q1_finish(r);
}

/* exact leg: order-independent correctly-rounded sums.
 * threads>1 parallelizes with OpenMP (exactness makes the split irrelevant);
 * group ids still reported in row-order first occurrence (computed serially,
 * cheap). Returns elapsed seconds for the baseline timing. */
EXPORT double o_q1_exact(int64_t n, const int32_t* shipdate, const double* qty,
                         const double* extprice, const double* discount, const double* tax,
                         const uint8_t* rflag, const uint8_t* lstatus,
                         int32_t cutoff, int32_t threads, o_q1_result* r)
{
    memset(r, 0, sizeof(*r));
    for (int c = 0; c < 6; c++) r->group_id_by_combo[c] = -1;
    const double S43 = 8796093022208.0;        /* 2^43 */
    const double S59 = 576460752303423488.0;   /* 2^59 */
    __int128 acc_base[6] = {0}, acc_dp[6] = {0}, acc_ch[6] = {0}, acc_disc[6] = {0};
    int64_t acc_qty[6] = {0}, cnt[6] = {0};
    double t0 = 0, t1 = 0;
#ifdef _OPENMP
    t0 = omp_get_wtime();
#endif
#ifdef _OPENMP
#pragma omp parallel num_threads(threads > 0 ? threads : 1)
#endif
    {
        __int128 l_base[6] = {0}, l_dp[6] = {0}, l_ch[6] = {0}, l_disc[6] = {0};
        int64_t l_qty[6] = {0}, l_cnt[6] = {0};
#ifdef _OPENMP
#pragma omp for schedule(static)
#endif
        for (int64_t i = 0; i < n; i++) {
            if (shipdate[i] > cutoff) continue;
            int c = rflag[i] * 2 + lstatus[i];
            double dp = extprice[i] * (1.0 - discount[i]);
            double ch = dp * (1.0 + tax[i]);
            l_base[c] += (__int128)(long long)(extprice[i] * S43);
            l_dp[c]   += (__int128)(long long)(dp * S43);
            l_ch[c]   += (__int128)(long long)(ch * S43);
            l_disc[c] += (__int128)(long long)(discount[i] * S59);
            l_qty[c]  += (long long)qty[i];
            l_cnt[c]++;
        }
#ifdef _OPENMP
#pragma omp critical
#endif
        for (int c = 0; c < 6; c++) {
            acc_base[c] += l_base[c]; acc_dp[c] += l_dp[c]; acc_ch[c] += l_ch[c];
            acc_disc[c] += l_disc[c]; acc_qty[c] += l_qty[c]; cnt[c] += l_cnt[c];
        }
    }
#ifdef _OPENMP
    t1 = omp_get_wtime();
#endif
    for (int c = 0; c < 6; c++) {
        r->sum_base[c] = (double)acc_base[c] / S43;
        r->sum_disc_price[c] = (double)acc_dp[c] / S43;
        r->sum_charge[c] = (double)acc_ch[c] / S43;
        r->sum_disc[c] = (double)acc_disc[c] / S59;
        r->sum_qty[c] = (double)acc_qty[c];
        r->count[c] = cnt[c];
    }
    /* row-order group ids */
    int32_t next_gid = 0;
    for (int64_t i = 0; i < n && next_gid < 6; i++) {
        if (shipdate[i] > cutoff) continue;
        int c = rflag[i] * 2 + lstatus[i];
        if (r->group_id_by_combo[c] < 0) r->group_id_by_combo[c] = next_gid++;
    }
    r->n_groups = next_gid;
    q1_finish(r);
    return t1 - t0;
}
