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
                               int64_t* tp_cents, int64_t* suppkey,
                               uint8_t* shipinstruct)
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
            if (shipinstruct)  shipinstruct[n] = l.shipinstruct;
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
                       cols->shipmode, cols->tp_cents, cols->suppkey,
                       cols->shipinstruct);
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
    /* pool-backed (was raw hipMalloc): every query regenerates ~20-40 GB
     * of columns, and the raw alloc/free churn between the 22 sweep
     * queries made the driver re-map fresh VRAM inside the NEXT query's
     * timed region (q15 measured 70 ms in-sweep vs 8.5 warm; the HIP
     * trace showed 40+ ms of host-side allocation gaps) */
    if (flags & 1) TG_POOL_ALLOC(s, &cols->orderkey, rows * 8);
    if (flags & 2) {
        TG_POOL_ALLOC(s, &cols->commitdate, rows * 4);
        TG_POOL_ALLOC(s, &cols->receiptdate, rows * 4);
    }
    if (flags & 4) TG_POOL_ALLOC(s, &cols->partkey, rows * 8);
    if (flags & 8) TG_POOL_ALLOC(s, &cols->shipmode, rows);
    if (flags & 16) TG_POOL_ALLOC(s, &cols->tp_cents, rows * 8);
    if (flags & 32) TG_POOL_ALLOC(s, &cols->suppkey, rows * 8);
    if (flags & 64) TG_POOL_ALLOC(s, &cols->shipinstruct, rows);
    TG_POOL_ALLOC(s, &cols->shipdate, rows * 4);
    TG_POOL_ALLOC(s, &cols->quantity, rows * 8);
    TG_POOL_ALLOC(s, &cols->extendedprice, rows * 8);
    TG_POOL_ALLOC(s, &cols->discount, rows * 8);
    TG_POOL_ALLOC(s, &cols->tax, rows * 8);
    TG_POOL_ALLOC(s, &cols->returnflag, rows);
    TG_POOL_ALLOC(s, &cols->linestatus, rows);
    return tg_tpch_gen_lineitem(s, sf, order_start, order_count, cols);
}

extern "C" tg_status tg_tpch_lineitem_free(tg_session* s, tg_tpch_lineitem_cols* cols)
{
    tg_pool_free(s, cols->orderkey);
    tg_pool_free(s, cols->commitdate);
    tg_pool_free(s, cols->receiptdate);
    tg_pool_free(s, cols->partkey);
    tg_pool_free(s, cols->shipmode);
    tg_pool_free(s, cols->tp_cents);
    tg_pool_free(s, cols->suppkey);
    tg_pool_free(s, cols->shipinstruct);
    tg_pool_free(s, cols->shipdate);
    tg_pool_free(s, cols->quantity);
    tg_pool_free(s, cols->extendedprice);
    tg_pool_free(s, cols->discount);
    tg_pool_free(s, cols->tax);
    tg_pool_free(s, cols->returnflag);
    tg_pool_free(s, cols->linestatus);
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

/* ================= round 2: extended tables + text columns =============== */
#include "../../oracle/tpch_text.h"

__global__ void k_gen_part2(int64_t part_start, int64_t part_count,
                            int64_t* partkey, int16_t* type_id,
                            uint8_t* brand, int32_t* size, uint8_t* container,
                            uint8_t* name_ids /* 5 per row */,
                            int64_t* retail_cents)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (part_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = part_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, part_start + part_count - first);
    tpch_rng ty, mf, br, sz, cn, nm;
    tpch_rng_init(&ty, TPCH_SEED_P_TYPE, 1); tpch_rng_skip(&ty, first - 1);
    tpch_rng_init(&mf, TPCH_SEED_P_MFG, 1);  tpch_rng_skip(&mf, first - 1);
    tpch_rng_init(&br, TPCH_SEED_P_BRND, 1); tpch_rng_skip(&br, first - 1);
    tpch_rng_init(&sz, TPCH_SEED_P_SIZE, 1); tpch_rng_skip(&sz, first - 1);
    tpch_rng_init(&cn, TPCH_SEED_P_CNTR, 1); tpch_rng_skip(&cn, first - 1);
    tpch_rng_init(&nm, TPCH_SEED_P_NAME, TPCH_P_NAME_USAGE);
    tpch_rng_skip(&nm, (first - 1) * TPCH_P_NAME_USAGE);
    for (int64_t i = 0; i < cnt; i++) {
        int64_t at = first - part_start + i;
        int64_t p = first + i;
        if (partkey) partkey[at] = p;
        int64_t tid = tpch_rng_int(&ty, 1, 150) - 1;
        if (type_id) type_id[at] = (int16_t)tid;
        int64_t m = tpch_rng_int(&mf, 1, 5);
        int64_t b = m * 10 + tpch_rng_int(&br, 1, 5);
        if (brand) brand[at] = (uint8_t)b;
        if (size) size[at] = (int32_t)tpch_rng_int(&sz, 1, 50);
        if (container) container[at] = (uint8_t)(tpch_rng_int(&cn, 1, 40) - 1);
        if (name_ids) tpch_part_name_ids(&nm, name_ids + at * 5);
        else { tpch_rng_skip(&nm, 5); nm.used += 5; }
        tpch_rng_row_finished(&nm);
        if (retail_cents) retail_cents[at] = tpch_part_price_cents(p);
    }
}

extern "C" tg_status tg_tpch_gen_part2(tg_session* s, double sf,
    int64_t part_start, int64_t part_count, int64_t* d_partkey,
    int16_t* d_type, uint8_t* d_brand, int32_t* d_size, uint8_t* d_container,
    uint8_t* d_name_ids, int64_t* d_retail_cents)
{
    (void)sf;
    int64_t n_groups = (part_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_part2, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       part_start, part_count, d_partkey, d_type, d_brand,
                       d_size, d_container, d_name_ids, d_retail_cents);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

/* partsupp: 4 rows per part in dbgen bridge order */
__global__ void k_gen_partsupp(double sf, int64_t part_start, int64_t part_count,
                               int64_t* partkey, int64_t* suppkey,
                               int32_t* availqty, int64_t* supplycost_cents)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (part_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = part_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, part_start + part_count - first);
    int64_t S = (int64_t)(10000 * sf);
    tpch_rng qt, sc;
    tpch_rng_init(&qt, TPCH_SEED_PS_QTY, 4);  tpch_rng_skip(&qt, (first - 1) * 4);
    tpch_rng_init(&sc, TPCH_SEED_PS_SCST, 4); tpch_rng_skip(&sc, (first - 1) * 4);
    for (int64_t i = 0; i < cnt; i++) {
        int64_t p = first + i;
        for (int j = 0; j < 4; j++) {
            int64_t at = (p - part_start) * 4 + j;
            if (partkey) partkey[at] = p;
            if (suppkey) suppkey[at] = (p + j * (S / 4 + (p - 1) / S)) % S + 1;
            int64_t q = tpch_rng_int(&qt, 1, 9999);
            if (availqty) availqty[at] = (int32_t)q;
            int64_t c = tpch_rng_int(&sc, 100, 100000);
            if (supplycost_cents) supplycost_cents[at] = c;
        }
        tpch_rng_row_finished(&qt);
        tpch_rng_row_finished(&sc);
    }
}

extern "C" tg_status tg_tpch_gen_partsupp(tg_session* s, double sf,
    int64_t part_start, int64_t part_count, int64_t* d_partkey,
    int64_t* d_suppkey, int32_t* d_availqty, int64_t* d_supplycost_cents)
{
    int64_t n_groups = (part_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_partsupp, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, part_start, part_count, d_partkey, d_suppkey,
                       d_availqty, d_supplycost_cents);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

__global__ void k_gen_supplier2(double sf, int64_t supp_start, int64_t supp_count,
                                int64_t* suppkey, uint8_t* nationkey,
                                int64_t* acctbal_cents)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (supp_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = supp_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, supp_start + supp_count - first);
    (void)sf;
    tpch_rng nk, ab;
    tpch_rng_init(&nk, TPCH_SEED_S_NKEY, 1); tpch_rng_skip(&nk, first - 1);
    tpch_rng_init(&ab, TPCH_SEED_S_ABAL, 1); tpch_rng_skip(&ab, first - 1);
    for (int64_t i = 0; i < cnt; i++) {
        int64_t at = first - supp_start + i;
        if (suppkey) suppkey[at] = first + i;
        int64_t v = tpch_rng_int(&nk, 0, 24);
        if (nationkey) nationkey[at] = (uint8_t)v;
        int64_t b = tpch_rng_int(&ab, -99999, 999999);
        if (acctbal_cents) acctbal_cents[at] = b;
        tpch_rng_row_finished(&nk);
        tpch_rng_row_finished(&ab);
    }
}

extern "C" tg_status tg_tpch_gen_supplier2(tg_session* s, double sf,
    int64_t supp_start, int64_t supp_count, int64_t* d_suppkey,
    uint8_t* d_nationkey, int64_t* d_acctbal_cents)
{
    int64_t n_groups = (supp_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_supplier2, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, supp_start, supp_count, d_suppkey, d_nationkey,
                       d_acctbal_cents);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

/* orders with derived status (0=F 1=O 2=P), totalprice cents, and o_comment
 * pool slices. Status/totalprice require walking the order's lines
 * (dbgen mk_order does the same). */
__global__ void k_gen_orders3(double sf, int64_t order_start, int64_t order_count,
                              int64_t* orderkey, int64_t* custkey,
                              int32_t* orderdate, uint8_t* priority,
                              uint8_t* orderstatus, int64_t* totalprice_cents,
                              int64_t* cmnt_off, int32_t* cmnt_len)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (order_count + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = order_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, order_start + order_count - first);
    tpch_order_streams s;
    tpch_order_streams_init(&s, sf);
    tpch_order_streams_seek(&s, first);
    tpch_rng cm;
    tpch_rng_init(&cm, TPCH_SEED_O_CMNT, 2);
    tpch_rng_skip(&cm, (first - 1) * 2);
    for (int64_t i = 0; i < cnt; i++) {
        tpch_order_row o;
        tpch_gen_order(&s, first + i, &o);
        int64_t at = first - order_start + i;
        if (orderkey)  orderkey[at] = o.orderkey;
        if (custkey)   custkey[at] = o.custkey;
        if (orderdate) orderdate[at] = o.orderdate;
        if (priority)  priority[at] = o.priority;
        if (orderstatus || totalprice_cents) {
            int64_t tp = 0;
            int nF = 0, nO = 0;
            for (int j = 0; j < o.line_count; j++) {
                tpch_lineitem_row l;
                tpch_gen_line(&s, &o, j, &l);
                tp += l.tp_cents;
                if (l.linestatus) nO++; else nF++;
            }
            if (totalprice_cents) totalprice_cents[at] = tp;
            if (orderstatus)
                orderstatus[at] = (nO == 0) ? 0 : (nF == 0 ? 1 : 2);
        }
        if (cmnt_off || cmnt_len) {
            int64_t off; int32_t len;
            tpch_text_slice(&cm, TPCH_CMNT_AVG_O, &off, &len);
            if (cmnt_off) cmnt_off[at] = off;
            if (cmnt_len) cmnt_len[at] = len;
        }
        tpch_rng_row_finished(&cm);
        tpch_order_row_finished(&s);
    }
}

extern "C" tg_status tg_tpch_gen_orders3(tg_session* s, double sf,
    int64_t order_start, int64_t order_count,
    int64_t* d_orderkey, int64_t* d_custkey, int32_t* d_orderdate,
    uint8_t* d_priority, uint8_t* d_orderstatus, int64_t* d_totalprice_cents,
    int64_t* d_cmnt_off, int32_t* d_cmnt_len)
{
    int64_t n_groups = (order_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_gen_orders3, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       sf, order_start, order_count, d_orderkey, d_custkey,
                       d_orderdate, d_priority, d_orderstatus,
                       d_totalprice_cents, d_cmnt_off, d_cmnt_len);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

/* ---- text pool on device (built host-side once, cached per-process) ---- */
static uint8_t* g_d_pool = nullptr;
static char* g_h_pool = nullptr;

const char* tg_tpch_host_pool(void)
{
    if (!g_h_pool) {
        g_h_pool = (char*)malloc(TPCH_TEXT_POOL_SIZE);
        tpch_text_pool_build(g_h_pool, TPCH_TEXT_POOL_SIZE);
    }
    return g_h_pool;
}

extern "C" tg_status tg_tpch_pool(tg_session* s, const uint8_t** d_pool)
{
    if (!g_d_pool) {
        const char* h = tg_tpch_host_pool();
        TG_HIP_CHECK(hipMalloc(&g_d_pool, TPCH_TEXT_POOL_SIZE));
        TG_HIP_CHECK(hipMemcpyAsync(g_d_pool, h, TPCH_TEXT_POOL_SIZE,
                                    hipMemcpyHostToDevice, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    }
    *d_pool = g_d_pool;
    return TG_OK;
}

/* LIKE '%seg1%seg2%...%' over pool slices: per row, find each literal
 * segment in order within pool[off, off+len). anchor_start/end pin the
 * first/last segment (LIKE without leading/trailing %%).
 * Mirrors io.trino.type.LikeFunctions / LikePattern matching semantics for
 * patterns without '_' wildcards. */
#define LIKE_MAX_SEGS 4
#define LIKE_MAX_PAT 64
struct LikePat {
    char bytes[LIKE_MAX_PAT];
    int seg_off[LIKE_MAX_SEGS], seg_len[LIKE_MAX_SEGS];
    int n_segs, anchor_start, anchor_end;
};

/* SWAR first-byte scan: next index >= from with s[i] == c, else -1
 * (byte-loop scanning measured 38.8 ms for Q13's 150M x ~49 B comments) */
__device__ static inline int like_find_byte(const uint8_t* s, int from,
                                            int len, uint8_t c)
{
    int i = from;
    for (; i < len && ((uintptr_t)(s + i) & 7); i++)
        if (s[i] == c) return i;
    const uint64_t pat = 0x0101010101010101ull * c;
    for (; i + 8 <= len; i += 8) {
        uint64_t w = *(const uint64_t*)(s + i) ^ pat;
        uint64_t m = (w - 0x0101010101010101ull) & ~w & 0x8080808080808080ull;
        if (m) return i + (__ffsll((unsigned long long)m) - 1) / 8;
    }
    for (; i < len; i++)
        if (s[i] == c) return i;
    return -1;
}

__device__ static bool like_match(const uint8_t* txt, int len, const LikePat& p)
{
    int pos = 0;
    for (int k = 0; k < p.n_segs; k++) {
        int sl = p.seg_len[k];
        const char* seg = p.bytes + p.seg_off[k];
        if (k == 0 && p.anchor_start) {
            if (len < sl) return false;
            for (int i = 0; i < sl; i++) if (txt[i] != (uint8_t)seg[i]) return false;
            pos = sl;
            continue;
        }
        if (k == p.n_segs - 1 && p.anchor_end) {
            if (len - pos < sl) return false;
            for (int i = 0; i < sl; i++)
                if (txt[len - sl + i] != (uint8_t)seg[i]) return false;
            pos = len;
            continue;
        }
        bool found = false;
        int i = pos;
        while (i + sl <= len) {
            i = like_find_byte(txt, i, len - sl + 1, (uint8_t)seg[0]);
            if (i < 0) break;
            bool eq = true;
            for (int j = 1; j < sl && eq; j++) eq = txt[i + j] == (uint8_t)seg[j];
            if (eq) { pos = i + sl; found = true; break; }
            i++;
        }
        if (!found) return false;
    }
    return true;
}

__global__ void k_pool_like(const uint8_t* __restrict__ pool,
                            const int64_t* __restrict__ offs,
                            const int32_t* __restrict__ lens, int64_t n,
                            LikePat p, uint8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        flags[i] = like_match(pool + offs[i], lens[i], p) ? 1 : 0;
}

__global__ void k_var_like(const uint8_t* __restrict__ bytes,
                           const int32_t* __restrict__ offsets, int64_t n,
                           LikePat p, uint8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        flags[i] = like_match(bytes + offsets[i], offsets[i + 1] - offsets[i], p)
                       ? 1 : 0;
}

/* ---- indexed pool LIKE -------------------------------------------------
 * Comment columns are SLICES of the one 300 MiB dbgen pool, so a floating
 * '%a%b%' pattern can be evaluated against the POOL once: find every
 * occurrence of each segment in the pool (two streaming scans + a u32
 * sort), then each row does one binary search per segment over the
 * LLC-resident position arrays — the DictionaryAwareColumnarFilter idea
 * (evaluate once per dictionary entry) applied to the text pool. The
 * per-row byte scan (like_match) measured 26.4 ms for Q13's 150M x ~49 B
 * comments at SF100; see profiles/ for the indexed number. */
tg_status run_sort_keys_u32(tg_session* s, uint32_t* d_keys, int64_t n);
tg_status run_sort_pairs_bits(tg_session* s, uint64_t* d_keys, int64_t* d_vals,
                              int64_t n, int bits);

struct LikeIdx {
    const uint32_t* occ[LIKE_MAX_SEGS];
    long long cnt[LIKE_MAX_SEGS];
    int slen[LIKE_MAX_SEGS];
    int n_segs;
};

__global__ void k_seg_occurrences(const uint8_t* __restrict__ pool,
                                  int64_t pool_len, LikePat p, int seg,
                                  uint32_t* __restrict__ out,
                                  int32_t* __restrict__ counter)
{
    const char* sg = p.bytes + p.seg_off[seg];
    int sl = p.seg_len[seg];
    int64_t last = pool_len - sl;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i <= last; i += stride) {
        if (pool[i] != (uint8_t)sg[0]) continue;
        bool eq = true;
        for (int j = 1; j < sl && eq; j++) eq = pool[i + j] == (uint8_t)sg[j];
        if (eq) {
            int32_t at = atomicAdd(counter, 1);
            if (out) out[at] = (uint32_t)i;
        }
    }
}

__device__ static inline long long lb_u32(const uint32_t* __restrict__ a,
                                          long long n, uint32_t v)
{
    long long lo = 0, hi = n;
    while (lo < hi) {
        long long m = (lo + hi) >> 1;
        if (a[m] < v) lo = m + 1;
        else hi = m;
    }
    return lo;
}

/* byte-scan in pool-offset order: rows sorted by slice offset visit the
 * pool sequentially (neighbouring rows read the same cache lines), then
 * the verdict scatters back by row id. The offset-random per-row scan
 * measured 26.4 ms for Q13's 150M x ~49 B slices; the binary-search
 * occurrence index measured ~51 ms (each 4 B probe drags a 64 B LLC
 * line); sorting first makes the scan itself cache-resident. */
__global__ void k_pool_like_sorted(const uint8_t* __restrict__ pool,
                                   const uint64_t* __restrict__ sorted_offs,
                                   const int64_t* __restrict__ len_row,
                                   int64_t n, LikePat p,
                                   uint8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int64_t lr = len_row[i];
        int32_t len = (int32_t)(lr >> 32);
        uint32_t row = (uint32_t)lr;
        flags[row] = like_match(pool + sorted_offs[i], len, p) ? 1 : 0;
    }
}

__global__ void k_pack_len_row(const int32_t* __restrict__ lens, int64_t n,
                               int64_t* __restrict__ len_row)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        len_row[i] = ((int64_t)lens[i] << 32) | (uint32_t)i;
}

__global__ void k_pool_like_indexed(const int64_t* __restrict__ offs,
                                    const int32_t* __restrict__ lens,
                                    int64_t n, LikeIdx ix,
                                    uint8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int64_t pos = offs[i];
        int64_t end = pos + lens[i];
        bool ok = true;
        for (int k = 0; k < ix.n_segs && ok; k++) {
            long long j = lb_u32(ix.occ[k], ix.cnt[k], (uint32_t)pos);
            /* greedy leftmost occurrence, same as like_match */
            if (j >= ix.cnt[k] || (int64_t)ix.occ[k][j] + ix.slen[k] > end)
                ok = false;
            else
                pos = (int64_t)ix.occ[k][j] + ix.slen[k];
        }
        flags[i] = ok ? 1 : 0;
    }
}

static tg_status parse_like(const char* pattern, LikePat* p)
{
    memset(p, 0, sizeof(*p));
    size_t L = strlen(pattern);
    if (L >= LIKE_MAX_PAT) { TG_SET_ERR("pattern too long"); return TG_ERR_UNSUPPORTED; }
    p->anchor_start = L > 0 && pattern[0] != '%';
    p->anchor_end = L > 0 && pattern[L - 1] != '%';
    int n = 0, bo = 0;
    const char* q = pattern;
    while (*q) {
        while (*q == '%') q++;
        if (!*q) break;
        const char* e = q;
        while (*e && *e != '%') {
            if (*e == '_') { TG_SET_ERR("'_' wildcard unsupported"); return TG_ERR_UNSUPPORTED; }
            e++;
        }
        if (n >= LIKE_MAX_SEGS) { TG_SET_ERR("too many %% segments"); return TG_ERR_UNSUPPORTED; }
        p->seg_off[n] = bo; p->seg_len[n] = (int)(e - q);
        memcpy(p->bytes + bo, q, e - q);
        bo += (int)(e - q);
        n++;
        q = e;
    }
    p->n_segs = n;
    return TG_OK;
}

/* flags[i] = (pool slice i LIKE pattern); slices device-resident */
extern "C" tg_status tg_pool_like_flags(tg_session* s, const int64_t* d_offs,
    const int32_t* d_lens, int64_t n, const char* pattern, uint8_t* d_flags)
{
    const uint8_t* d_pool = nullptr;
    tg_status st = tg_tpch_pool(s, &d_pool);
    if (st != TG_OK) return st;
    LikePat p;
    st = parse_like(pattern, &p);
    if (st != TG_OK) return st;
    /* TG_LIKE_IDX: 0 = per-row byte scan, 1 = occurrence index,
     * 2 = offset-sorted byte scan (default above 1M rows — A/B in
     * profiles/: plain 26.4 ms / indexed ~51 / sorted see r02) */
    const char* ev = getenv("TG_LIKE_IDX");
    char mode = ev ? ev[0] : ((n >= (1 << 20)) ? '2' : '0');
    if (mode == '2') {
        uint64_t* d_so = nullptr;
        int64_t* d_lr = nullptr;
        TG_POOL_ALLOC(s, &d_so, n * 8);
        TG_POOL_ALLOC(s, &d_lr, n * 8);
        TG_HIP_CHECK(hipMemcpyAsync(d_so, d_offs, n * 8,
                                    hipMemcpyDeviceToDevice, s->stream));
        hipLaunchKernelGGL(k_pack_len_row, dim3(tg_grid_for(n)), dim3(TG_BLOCK),
                           0, s->stream, d_lens, n, d_lr);
        TG_HIP_CHECK(hipGetLastError());
        tg_status sst = run_sort_pairs_bits(s, d_so, d_lr, n, 29);
        if (sst != TG_OK) return sst;
        hipLaunchKernelGGL(k_pool_like_sorted, dim3(tg_grid_for(n)),
                           dim3(TG_BLOCK), 0, s->stream, d_pool, d_so, d_lr,
                           n, p, d_flags);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_so);
        tg_pool_free(s, d_lr);
        return TG_OK;
    }
    if (mode == '1' && !p.anchor_start && !p.anchor_end && p.n_segs >= 1) {
        /* floating pattern over many pool slices: index the pool once */
        LikeIdx ix{};
        ix.n_segs = p.n_segs;
        int32_t* d_cnt = nullptr;
        TG_POOL_ALLOC(s, &d_cnt, 4);
        uint32_t* bufs[LIKE_MAX_SEGS] = {};
        tg_status ist = TG_OK;
        for (int k = 0; k < p.n_segs && ist == TG_OK; k++) {
            TG_HIP_CHECK(hipMemsetAsync(d_cnt, 0, 4, s->stream));
            hipLaunchKernelGGL(k_seg_occurrences,
                               dim3(tg_grid_for(TPCH_TEXT_POOL_SIZE)),
                               dim3(TG_BLOCK), 0, s->stream, d_pool,
                               (int64_t)TPCH_TEXT_POOL_SIZE, p, k,
                               (uint32_t*)nullptr, d_cnt);
            TG_HIP_CHECK(hipGetLastError());
            int32_t cnt = 0;
            TG_HIP_CHECK(hipMemcpy(&cnt, d_cnt, 4, hipMemcpyDeviceToHost));
            ist = tg_pool_alloc(s, (void**)&bufs[k],
                                (int64_t)(cnt ? cnt : 1) * 4);
            if (ist != TG_OK) break;
            TG_HIP_CHECK(hipMemsetAsync(d_cnt, 0, 4, s->stream));
            hipLaunchKernelGGL(k_seg_occurrences,
                               dim3(tg_grid_for(TPCH_TEXT_POOL_SIZE)),
                               dim3(TG_BLOCK), 0, s->stream, d_pool,
                               (int64_t)TPCH_TEXT_POOL_SIZE, p, k, bufs[k],
                               d_cnt);
            TG_HIP_CHECK(hipGetLastError());
            ist = run_sort_keys_u32(s, bufs[k], cnt);
            ix.occ[k] = bufs[k];
            ix.cnt[k] = cnt;
            ix.slen[k] = p.seg_len[k];
        }
        if (ist == TG_OK) {
            hipLaunchKernelGGL(k_pool_like_indexed, dim3(tg_grid_for(n)),
                               dim3(TG_BLOCK), 0, s->stream, d_offs, d_lens,
                               n, ix, d_flags);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        }
        for (int k = 0; k < p.n_segs; k++)
            if (bufs[k]) tg_pool_free(s, bufs[k]);
        tg_pool_free(s, d_cnt);
        return ist;
    }
    hipLaunchKernelGGL(k_pool_like, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0,
                       s->stream, d_pool, d_offs, d_lens, n, p, d_flags);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

/* generic VARCHAR LIKE on a device var-width column (bytes + offsets[n+1]) */
extern "C" tg_status tg_varchar_like_flags(tg_session* s, const uint8_t* d_bytes,
    const int32_t* d_offsets, int64_t n, const char* pattern, uint8_t* d_flags)
{
    LikePat p;
    tg_status st = parse_like(pattern, &p);
    if (st != TG_OK) return st;
    hipLaunchKernelGGL(k_var_like, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0,
                       s->stream, d_bytes, d_offsets, n, p, d_flags);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

/* supplier comments as a device var-width column, WITH the dbgen BBB
 * overlay ("Customer ...Complaints"/"Recommends" written over 10 rows per
 * 10,000; selection/type/junk/offset each one independent stream draw/row).
 * Overlay semantics pinned by the SF1 Q16 answer fixture. */
__global__ void k_s_cmnt_lens(int64_t supp_start, int64_t n, int32_t* lens)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (n + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = supp_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, supp_start + n - first);
    tpch_rng cm;
    tpch_rng_init(&cm, TPCH_SEED_S_CMNT, 2);
    tpch_rng_skip(&cm, (first - 1) * 2);
    for (int64_t i = 0; i < cnt; i++) {
        int64_t off; int32_t len;
        tpch_text_slice(&cm, TPCH_CMNT_AVG_S, &off, &len);
        lens[first - supp_start + i] = len;
        tpch_rng_row_finished(&cm);
    }
}

__global__ void k_s_cmnt_bytes(const uint8_t* __restrict__ pool,
                               int64_t supp_start, int64_t n,
                               const int32_t* __restrict__ offsets,
                               uint8_t* __restrict__ out)
{
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n_groups = (n + GOT - 1) / GOT;
    if (g >= n_groups) return;
    int64_t first = supp_start + g * GOT;
    int64_t cnt = min((int64_t)GOT, supp_start + n - first);
    tpch_rng cm, bs, bj, bo, bt;
    tpch_rng_init(&cm, TPCH_SEED_S_CMNT, 2);
    tpch_rng_skip(&cm, (first - 1) * 2);
    tpch_rng_init(&bs, TPCH_SEED_BBB_CMNT, 1); tpch_rng_skip(&bs, first - 1);
    tpch_rng_init(&bj, TPCH_SEED_BBB_JNK, 1);  tpch_rng_skip(&bj, first - 1);
    tpch_rng_init(&bo, TPCH_SEED_BBB_OFF, 1);  tpch_rng_skip(&bo, first - 1);
    tpch_rng_init(&bt, TPCH_SEED_BBB_TYPE, 1); tpch_rng_skip(&bt, first - 1);
    const char* BASE = "Customer ";
    const char* GOOD = "Recommends";
    const char* BAD = "Complaints";
    for (int64_t i = 0; i < cnt; i++) {
        int64_t off; int32_t len;
        tpch_text_slice(&cm, TPCH_CMNT_AVG_S, &off, &len);
        int64_t at = first - supp_start + i;
        uint8_t* dst = out + offsets[at];
        for (int32_t b = 0; b < len; b++) dst[b] = pool[off + b];
        int64_t sel = tpch_rng_int(&bs, 1, 10000);
        int64_t type = tpch_rng_int(&bt, 0, 100);
        int64_t junk = tpch_rng_int(&bj, 0, len - 19);
        int64_t boff = tpch_rng_int(&bo, 0, len - (19 + junk));
        if (sel <= 10) {
            const char* tail = (type < 50) ? BAD : GOOD;
            for (int b = 0; b < 9; b++) dst[boff + b] = (uint8_t)BASE[b];
            for (int b = 0; b < 10; b++) dst[boff + 9 + junk + b] = (uint8_t)tail[b];
        }
        tpch_rng_row_finished(&cm);
        tpch_rng_row_finished(&bs);
        tpch_rng_row_finished(&bj);
        tpch_rng_row_finished(&bo);
        tpch_rng_row_finished(&bt);
    }
}

tg_status run_scan_i32(tg_session* s, int32_t* d_arr, int64_t n, int32_t* d_total);

extern "C" tg_status tg_tpch_gen_supplier_comments(tg_session* s, double sf,
    int64_t supp_start, int64_t supp_count,
    int32_t** d_offsets_out /* [n+1], pool-allocated */,
    uint8_t** d_bytes_out)
{
    (void)sf;
    const uint8_t* d_pool = nullptr;
    tg_status st = tg_tpch_pool(s, &d_pool);
    if (st != TG_OK) return st;
    int32_t* d_off = nullptr;
    TG_POOL_ALLOC(s, &d_off, (supp_count + 1) * 4);
    int64_t n_groups = (supp_count + GOT - 1) / GOT;
    int grid = (int)((n_groups + TG_BLOCK - 1) / TG_BLOCK);
    hipLaunchKernelGGL(k_s_cmnt_lens, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       supp_start, supp_count, d_off);
    TG_HIP_CHECK(hipGetLastError());
    int32_t* d_total = nullptr;
    TG_POOL_ALLOC(s, &d_total, 4);
    st = run_scan_i32(s, d_off, supp_count, d_total);
    if (st != TG_OK) return st;
    int32_t total = 0;
    TG_HIP_CHECK(hipMemcpyAsync(&total, d_total, 4, hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    TG_HIP_CHECK(hipMemcpyAsync(d_off + supp_count, d_total, 4,
                                hipMemcpyDeviceToDevice, s->stream));
    uint8_t* d_bytes = nullptr;
    TG_POOL_ALLOC(s, &d_bytes, total ? total : 1);
    hipLaunchKernelGGL(k_s_cmnt_bytes, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       d_pool, supp_start, supp_count, d_off, d_bytes);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_total);
    *d_offsets_out = d_off;
    *d_bytes_out = d_bytes;
    return TG_OK;
}

/* p_name predicate helper: flags[i] = name contains color `cid` (any of the
 * 5 words) or, with first_only, name starts with it ('forest%'-class
 * prefixes / '%green%'-class contains — color words never occur as
 * substrings of other colors at word boundaries in the 92-color dist). */
__global__ void k_part_name_flag(const uint8_t* __restrict__ name_ids,
                                 int64_t n, int cid, int first_only,
                                 uint8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        const uint8_t* w = name_ids + i * 5;
        bool f = first_only ? (w[0] == cid)
                            : (w[0] == cid || w[1] == cid || w[2] == cid ||
                               w[3] == cid || w[4] == cid);
        flags[i] = f ? 1 : 0;
    }
}

extern "C" tg_status tg_tpch_part_name_flag(tg_session* s,
    const uint8_t* d_name_ids, int64_t n, int32_t color_id, int32_t first_only,
    uint8_t* d_flags)
{
    hipLaunchKernelGGL(k_part_name_flag, dim3(tg_grid_for(n)), dim3(TG_BLOCK),
                       0, s->stream, d_name_ids, n, color_id, first_only,
                       d_flags);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}
