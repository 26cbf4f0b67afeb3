/* tpchgen.hip — device-side TPC-H lineitem/orders/customer column generator
 * for gfx950. Bench/test INPUT infrastructure (synthetic data maker), not the
 * measured compute path: bench.py's timed region never includes generation.
 * Includes oracle/tpch_core.h (header-only) so device and CPU-oracle streams
 * are bit-identical by construction; no oracle object code is linked.
 *
 * Strategy: one thread owns TPCH_GEN_ORDERS_PER_THREAD consecutive orders.
 * Pass 1: per-thread line counts (O_LCNT stream only) -> host exclusive scan
 * (cheap: orders/64 entries). Pass 2: per-thread log-time seek of all streams
 * then sequential generation, rows written at the scanned offsets.
 */
#include "common.h"

#define TPCH_HD __host__ __device__
#include "../../oracle/tpch_core.h"

constexpr int GOT = 64;   /* orders per thread */

__global__ void k_lcnt_counts(double sf, int64_t order_start, int64_t order_count,
                              int64_t* group_counts, int64_t n_groups)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    int64_t first = order_start + g * GOT;
    int64_t cnt_orders = min((int64_t)GOT, order_start + order_count - first);
    tpch_rng lcnt;
    tpch_rng_init(&lcnt, TPCH_SEED_O_LCNT, 1);
    tpch_rng_skip(&lcnt, first - 1);
    int64_t total = 0;
    for (int64_t i = 0; i < cnt_orders; i++)
        total += tpch_rng_int(&lcnt, 1, TPCH_LINES_PER_ORDER_MAX);
    group_counts[g] = total;
}

__global__ void k_gen_lineitem(double sf, int64_t order_start, int64_t order_count,
                               const int64_t* group_offsets, int64_t n_groups,
                               int64_t* orderkey, int32_t* shipdate,
                               double* quantity, double* extendedprice,
                               double* discount, double* tax,
                               uint8_t* returnflag, uint8_t* linestatus,
                               int32_t* commitdate, int32_t* receiptdate,
                               int64_t* partkey, uint8_t* shipmode,
                               int64_t* tp_cents, int64_t* suppkey)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= n_groups) return;
    int64_t first = order_start + g * GOT;
    int64_t cnt_orders = min((int64_t)GOT, order_start + order_count - first);
    tpch_order_streams s;
    tpch_order_streams_init(&s, sf);
    tpch_order_streams_seek(&s, first);
    int64_t n = group_offsets[g];
    for (int64_t i = 0; i < cnt_orders; i++) {
        tpch_order_row o;
        tpch_gen_order(&s, first + i, &o);
        for (int j = 0; j < o.line_count; j++, n++) {
            tpch_lineitem_row l;
            tpch_gen_line(&s, &o, j, &l);
            if (orderkey)      orderkey[n] = l.orderkey;
            if (shipdate)      shipdate[n] = l.shipdate;
            if (quantity)      quantity[n] = (double)l.qty;
            if (extendedprice) extendedprice[n] = tpch_cents_to_double(l.extprice_cents);
            if (discount)      discount[n] = (double)l.discount_pct / 100.0;
            if (tax)           tax[n] = (double)l.tax_pct / 100.0;
            if (returnflag)    returnflag[n] = l.returnflag;
            if (linestatus)    linestatus[n] = l.linestatus;
            if (commitdate)    commitdate[n] = l.commitdate;
            if (receiptdate)   receiptdate[n] = l.receiptdate;
            if (partkey)       partkey[n] = l.partkey;
            if (shipmode)      shipmode[n] = l.shipmode;
            if (tp_cents)      tp_cents[n] = l.tp_cents;
            if (suppkey)       suppkey[n] = l.suppkey;
        }
        tpch_order_row_finished(&s);
    }
}

__global__ void k_gen_orders(double sf, int64_t order_start, int64_t order_count,
                             int64_t* orderkey, int64_t* custkey, int32_t* orderdate,
                             uint8_t* priority)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (order_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = order_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, order_start + order_count - first);
    tpch_order_streams s;
    tpch_order_streams_init(&s, sf);
    tpch_order_streams_seek(&s, first);
    for (int64_t i = 0; i < cnt; i++) {
        tpch_order_row o;
        tpch_gen_order(&s, first + i, &o);
        int64_t at = first - order_start + i;
        if (orderkey)  orderkey[at] = o.orderkey;
        if (custkey)   custkey[at] = o.custkey;
        if (orderdate) orderdate[at] = o.orderdate;
        if (priority)  priority[at] = o.priority;
        tpch_order_row_finished(&s);
    }
}

__global__ void k_gen_supplier(double sf, int64_t supp_start, int64_t supp_count,
                               int64_t* suppkey, uint8_t* nationkey)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (supp_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = supp_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, supp_start + supp_count - first);
    tpch_rng nk;
    tpch_rng_init(&nk, TPCH_SEED_S_NKEY, 1);
    tpch_rng_skip(&nk, first - 1);
    for (int64_t i = 0; i < cnt; i++) {
        int64_t at = first - supp_start + i;
        if (suppkey) suppkey[at] = first + i;
        int64_t v = tpch_rng_int(&nk, 0, 24);
        if (nationkey) nationkey[at] = (uint8_t)v;
        tpch_rng_row_finished(&nk);
    }
}

__global__ void k_gen_part(int64_t part_start, int64_t part_count,
                           int64_t* partkey, int16_t* type_id)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (part_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = part_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, part_start + part_count - first);
    uint8_t tmp[GOT];
    tpch_gen_part(first, cnt, partkey ? partkey + (first - part_start) : nullptr,
                  type_id ? tmp : nullptr);
    if (type_id)
        for (int64_t i = 0; i < cnt; i++) type_id[first - part_start + i] = tmp[i];
}

__global__ void k_gen_customer(double sf, int64_t cust_start, int64_t cust_count,
                               int64_t* custkey, uint8_t* mktsegment,
                               uint8_t* nationkey, int64_t* acctbal_cents)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (cust_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = cust_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, cust_start + cust_count - first);
    tpch_rng mseg, nk, abal;
    tpch_rng_init(&mseg, TPCH_SEED_C_MSEG, 1);
    tpch_rng_init(&nk,   TPCH_SEED_C_NKEY, 1);
    tpch_rng_init(&abal, TPCH_SEED_C_ABAL, 1);
    tpch_rng_skip(&mseg, first - 1);
    tpch_rng_skip(&nk,   first - 1);
    tpch_rng_skip(&abal, first - 1);
    for (int64_t i = 0; i < cnt; i++) {
        int64_t at = first - cust_start + i;
        if (custkey) custkey[at] = first + i;
        int64_t pick = tpch_rng_int(&mseg, 1, 5);
        if (mktsegment) mktsegment[at] = (uint8_t)(pick - 1);
        int64_t nkv = tpch_rng_int(&nk, 0, 24);
        if (nationkey) nationkey[at] = (uint8_t)nkv;
        int64_t bal = tpch_rng_int(&abal, -99999, 999999);
        if (acctbal_cents) acctbal_cents[at] = bal;
        tpch_rng_row_finished(&mseg);
        tpch_rng_row_finished(&nk);
        tpch_rng_row_finished(&abal);
    }
}

/* host: count rows for an order range (pass 1 + host scan). offsets_out
 * (optional, device, n_groups entries) receives the exclusive scan. */
extern "C" tg_status tg_tpch_lineitem_rows(tg_session* s, double sf,
    int64_t order_start, int64_t order_count, int64_t* row_count_out,
    int64_t** dev_offsets_out /* optional; caller frees with hipFree */)
{
    int64_t n_groups = (order_count + GOT - 1) / GOT;
    int64_t* d_counts = nullptr;
    TG_HIP_CHECK(hipMalloc(&d_counts, n_groups * sizeof(int64_t)));
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_lcnt_counts, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, order_start, order_count, d_counts, n_groups);
    TG_HIP_CHECK(hipGetLastError());
    int64_t* h_counts = (int64_t*)malloc(n_groups * sizeof(int64_t));
    TG_HIP_CHECK(hipMemcpyAsync(h_counts, d_counts, n_groups * sizeof(int64_t),
                                hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    int64_t total = 0;
    for (int64_t i = 0; i < n_groups; i++) {   /* exclusive scan in place */
        int64_t c = h_counts[i];
        h_counts[i] = total;
        total += c;
    }
    *row_count_out = total;
    if (dev_offsets_out) {
        TG_HIP_CHECK(hipMemcpyAsync(d_counts, h_counts, n_groups * sizeof(int64_t),
                                    hipMemcpyHostToDevice, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        *dev_offsets_out = d_counts;
    }
    else {
        TG_HIP_CHECK(hipFree(d_counts));
    }
    free(h_counts);
    return TG_OK;
}

extern "C" tg_status tg_tpch_gen_lineitem(tg_session* s, double sf,
    int64_t order_start, int64_t order_count, tg_tpch_lineitem_cols* cols)
{
    int64_t rows = 0;
    int64_t* d_offsets = nullptr;
    tg_status st = tg_tpch_lineitem_rows(s, sf, order_start, order_count, &rows, &d_offsets);
    if (st != TG_OK) return st;
    int64_t n_groups = (order_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_lineitem, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, order_start, order_count, d_offsets, n_groups,
                       cols->orderkey, cols->shipdate, cols->quantity,
                       cols->extendedprice, cols->discount, cols->tax,
                       cols->returnflag, cols->linestatus,
                       cols->commitdate, cols->receiptdate, cols->partkey,
                       cols->shipmode, cols->tp_cents, cols->suppkey);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    TG_HIP_CHECK(hipFree(d_offsets));
    cols->row_count = rows;
    return TG_OK;
}

/* allocate-and-generate convenience (exact sizing; used by bench/tests) */
extern "C" tg_status tg_tpch_lineitem_alloc(tg_session* s, double sf,
    int64_t order_start, int64_t order_count, int flags,
    tg_tpch_lineitem_cols* cols)
{
    memset(cols, 0, sizeof(*cols));
    int64_t rows = 0;
    tg_status st = tg_tpch_lineitem_rows(s, sf, order_start, order_count, &rows, nullptr);
    if (st != TG_OK) return st;
    if (flags & 1) TG_HIP_CHECK(hipMalloc(&cols->orderkey, rows * 8));
    if (flags & 2) {
        TG_HIP_CHECK(hipMalloc(&cols->commitdate, rows * 4));
        TG_HIP_CHECK(hipMalloc(&cols->receiptdate, rows * 4));
    }
    if (flags & 4) TG_HIP_CHECK(hipMalloc(&cols->partkey, rows * 8));
    if (flags & 8) TG_HIP_CHECK(hipMalloc(&cols->shipmode, rows));
    if (flags & 16) TG_HIP_CHECK(hipMalloc(&cols->tp_cents, rows * 8));
    if (flags & 32) TG_HIP_CHECK(hipMalloc(&cols->suppkey, rows * 8));
    TG_HIP_CHECK(hipMalloc(&cols->shipdate, rows * 4));
    TG_HIP_CHECK(hipMalloc(&cols->quantity, rows * 8));
    TG_HIP_CHECK(hipMalloc(&cols->extendedprice, rows * 8));
    TG_HIP_CHECK(hipMalloc(&cols->discount, rows * 8));
    TG_HIP_CHECK(hipMalloc(&cols->tax, rows * 8));
    TG_HIP_CHECK(hipMalloc(&cols->returnflag, rows));
    TG_HIP_CHECK(hipMalloc(&cols->linestatus, rows));
    return tg_tpch_gen_lineitem(s, sf, order_start, order_count, cols);
}

extern "C" tg_status tg_tpch_lineitem_free(tg_session* s, tg_tpch_lineitem_cols* cols)
{
    (void)s;
    if (cols->orderkey) TG_HIP_CHECK(hipFree(cols->orderkey));
    if (cols->commitdate) TG_HIP_CHECK(hipFree(cols->commitdate));
    if (cols->receiptdate) TG_HIP_CHECK(hipFree(cols->receiptdate));
    if (cols->partkey) TG_HIP_CHECK(hipFree(cols->partkey));
    if (cols->shipmode) TG_HIP_CHECK(hipFree(cols->shipmode));
    if (cols->tp_cents) TG_HIP_CHECK(hipFree(cols->tp_cents));
    if (cols->suppkey) TG_HIP_CHECK(hipFree(cols->suppkey));
    TG_HIP_CHECK(hipFree(cols->shipdate));
    TG_HIP_CHECK(hipFree(cols->quantity));
    TG_HIP_CHECK(hipFree(cols->extendedprice));
    TG_HIP_CHECK(hipFree(cols->discount));
    TG_HIP_CHECK(hipFree(cols->tax));
    TG_HIP_CHECK(hipFree(cols->returnflag));
    TG_HIP_CHECK(hipFree(cols->linestatus));
    memset(cols, 0, sizeof(*cols));
    return TG_OK;
}

extern "C" tg_status tg_tpch_gen_orders(tg_session* s, double sf,
    int64_t order_start, int64_t order_count,
    int64_t* d_orderkey, int64_t* d_custkey, int32_t* d_orderdate,
    uint8_t* d_priority)
{
    int64_t n_groups = (order_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_orders, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, order_start, order_count, d_orderkey, d_custkey, d_orderdate,
                       d_priority);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

extern "C" tg_status tg_tpch_gen_customer(tg_session* s, double sf,
    int64_t cust_start, int64_t cust_count, int64_t* d_custkey, uint8_t* d_mktsegment,
    uint8_t* d_nationkey, int64_t* d_acctbal_cents)
{
    int64_t n_groups = (cust_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_customer, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, cust_start, cust_count, d_custkey, d_mktsegment,
                       d_nationkey, d_acctbal_cents);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

extern "C" tg_status tg_tpch_gen_supplier(tg_session* s, double sf,
    int64_t supp_start, int64_t supp_count, int64_t* d_suppkey, uint8_t* d_nationkey)
{
    int64_t n_groups = (supp_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_supplier, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, supp_start, supp_count, d_suppkey, d_nationkey);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

extern "C" tg_status tg_tpch_gen_part(tg_session* s, double sf,
    int64_t part_start, int64_t part_count, int64_t* d_partkey, int16_t* d_type)
{
    (void)sf;
    int64_t n_groups = (part_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_part, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       part_start, part_count, d_partkey, d_type);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}
