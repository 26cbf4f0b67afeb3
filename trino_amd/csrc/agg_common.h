/* agg_common.h — aggregation state + kernels shared by the hash
 * (ops_groupby.hip) and streaming (ops_streamagg.hip) aggregation
 * operators. Definitions live in ops_groupby.hip. */
#pragma once
#include "dev_hash.h"

struct KAgg {
    int32_t fn;           /* tg_agg_fn */
    int32_t in_ch;        /* -1 for COUNT_STAR; AVG FINAL: count ch, sum = ch+1;
                             SUM_F64_EXACT FINAL: lo ch, hi = ch+1 */
    double* sum;          /* f64 state (or null); SUM_F64_EXACT: i128 HI words */
    long long* cnt;       /* i64 state (count / int sum); SUM_F64_EXACT: LO words */
    double scale;         /* SUM_F64_EXACT: 2^scale_pow */
    int32_t mask_a = -1;  /* row gate col[mask_a] > col[mask_b]; a==b => off */
    int32_t mask_b = -1;
};
#define MAX_AGGS 12

__global__ void k_agg_update(const int32_t* gids, int64_t n,
                             const KColH* cols, const KAgg* aggs, int n_aggs,
                             int step);
__global__ void k_agg_update_sorted(const int32_t* gids, int64_t n,
                                    const KColH* cols, const KAgg* aggs,
                                    int n_aggs, int step);
__global__ void k_emit_f64(const double* state, const int32_t* old_by_new,
                           int32_t n, double* out);
__global__ void k_emit_i64(const long long* state, const int32_t* old_by_new,
                           int32_t n, int64_t* out);
__global__ void k_emit_i64_biased(const long long* state,
                                  const int32_t* old_by_new, int32_t n,
                                  int64_t* out);
__global__ void k_emit_avg(const double* sum, const long long* cnt,
                           const int32_t* old_by_new, int32_t n, double* out,
                           uint64_t* out_valid);
__global__ void k_emit_exact(const unsigned long long* lo,
                             const unsigned long long* hi,
                             const int32_t* old_by_new, int32_t n,
                             double inv_scale, double* out);
