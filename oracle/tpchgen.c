/* tpchgen.c — CPU oracle entry points for the TPC-H generator restatement.
 * ORACLE / TEST INFRASTRUCTURE ONLY (see tpch_core.h header for provenance
 * and parity pins). Exported for ctypes use by tests/ and bench.py's
 * cpu_baseline leg only; the product path never links this.
 */
#include "tpch_core.h"
#include <string.h>

int64_t tpch_gen_orders2(double sf, int64_t order_start, int64_t order_count,
    int64_t* orderkey, int64_t* custkey, int32_t* orderdate, uint8_t* priority);

#define EXPORT __attribute__((visibility("default")))

/* Count lineitem rows for a dense order range [order_start, order_start+order_count)
 * (1-based). Only consumes the O_LCNT stream. */
EXPORT int64_t tpch_lineitem_count(double sf, int64_t order_start, int64_t order_count)
{
    (void)sf;
    tpch_rng lcnt;
    tpch_rng_init(&lcnt, TPCH_SEED_O_LCNT, 1);
    tpch_rng_skip(&lcnt, order_start - 1);
    int64_t total = 0;
    for (int64_t i = 0; i < order_count; i++)
        total += tpch_rng_int(&lcnt, 1, TPCH_LINES_PER_ORDER_MAX);
    return total;
}

/* Generate lineitem columns for a dense order range. Output arrays must hold
 * >= 7*order_count entries; any pointer may be NULL to skip that column.
 * Doubles are exactly the values the reference's cursor surfaces
 * (TpchRecordSet.java:192-210: DATE as epoch-day long, DOUBLE = cents/100.0).
 * Returns the number of rows written. */
EXPORT int64_t tpch_gen_lineitem(double sf, int64_t order_start, int64_t order_count,
    int64_t* orderkey, int64_t* partkey, int32_t* linenumber,
    int32_t* shipdate, int32_t* commitdate, int32_t* receiptdate,
    double* quantity, double* extendedprice, double* discount, double* tax,
    uint8_t* returnflag, uint8_t* linestatus, uint8_t* shipmode,
    int64_t* tp_cents, int64_t* suppkey)
{
    tpch_order_streams s;
    tpch_order_streams_init(&s, sf);
    tpch_order_streams_seek(&s, order_start);
    int64_t n = 0;
    for (int64_t i = 0; i < order_count; i++) {
        tpch_order_row o;
        tpch_gen_order(&s, order_start + i, &o);
        for (int j = 0; j < o.line_count; j++, n++) {
            tpch_lineitem_row l;
            tpch_gen_line(&s, &o, j, &l);
            if (orderkey)      orderkey[n] = l.orderkey;
            if (partkey)       partkey[n] = l.partkey;
            if (linenumber)    linenumber[n] = l.linenumber;
            if (shipdate)      shipdate[n] = l.shipdate;
            if (commitdate)    commitdate[n] = l.commitdate;
            if (receiptdate)   receiptdate[n] = l.receiptdate;
            if (quantity)      quantity[n] = (double)l.qty;
            if (extendedprice) extendedprice[n] = tpch_cents_to_double(l.extprice_cents);
            if (discount)      discount[n] = (double)l.discount_pct / 100.0;
            if (tax)           tax[n] = (double)l.tax_pct / 100.0;
            if (returnflag)    returnflag[n] = l.returnflag;
            if (linestatus)    linestatus[n] = l.linestatus;
            if (shipmode)      shipmode[n] = l.shipmode;
            if (tp_cents)      tp_cents[n] = l.tp_cents;
            if (suppkey)       suppkey[n] = l.suppkey;
        }
        tpch_order_row_finished(&s);
    }
    return n;
}

/* Generate order headers (Q3 build side): orderkey, custkey, orderdate. */
EXPORT int64_t tpch_gen_orders(double sf, int64_t order_start, int64_t order_count,
    int64_t* orderkey, int64_t* custkey, int32_t* orderdate)
{
    return tpch_gen_orders2(sf, order_start, order_count, orderkey, custkey,
                            orderdate, 0);
}

EXPORT int64_t tpch_gen_orders2(double sf, int64_t order_start, int64_t order_count,
    int64_t* orderkey, int64_t* custkey, int32_t* orderdate, uint8_t* priority)
{
    tpch_order_streams s;
    tpch_order_streams_init(&s, sf);
    tpch_order_streams_seek(&s, order_start);
    for (int64_t i = 0; i < order_count; i++) {
        tpch_order_row o;
        tpch_gen_order(&s, order_start + i, &o);
        if (orderkey)  orderkey[i] = o.orderkey;
        if (custkey)   custkey[i] = o.custkey;
        if (orderdate) orderdate[i] = o.orderdate;
        if (priority)  priority[i] = o.priority;
        tpch_order_row_finished(&s);
    }
    return order_count;
}

/* Customer mktsegment ids (Q3): id 0..4 in dists.dss order
 * AUTOMOBILE,BUILDING,FURNITURE,MACHINERY,HOUSEHOLD. custkey = dense index. */
EXPORT int64_t tpch_gen_customer(double sf, int64_t cust_start, int64_t cust_count,
    int64_t* custkey, uint8_t* mktsegment, uint8_t* nationkey,
    int64_t* acctbal_cents)
{
    (void)sf;
    tpch_rng mseg, nk, abal;
    tpch_rng_init(&mseg, TPCH_SEED_C_MSEG, 1);
    tpch_rng_init(&nk,   TPCH_SEED_C_NKEY, 1);
    tpch_rng_init(&abal, TPCH_SEED_C_ABAL, 1);
    tpch_rng_skip(&mseg, cust_start - 1);
    tpch_rng_skip(&nk,   cust_start - 1);
    tpch_rng_skip(&abal, cust_start - 1);
    for (int64_t i = 0; i < cust_count; i++) {
        if (custkey) custkey[i] = cust_start + i;
        int64_t pick = tpch_rng_int(&mseg, 1, 5);
        if (mktsegment) mktsegment[i] = (uint8_t)(pick - 1);
        int64_t nkv = tpch_rng_int(&nk, 0, 24);
        if (nationkey) nationkey[i] = (uint8_t)nkv;
        int64_t bal = tpch_rng_int(&abal, -99999, 999999);
        if (acctbal_cents) acctbal_cents[i] = bal;
        tpch_rng_row_finished(&mseg);
        tpch_rng_row_finished(&nk);
        tpch_rng_row_finished(&abal);
    }
    return cust_count;
}

EXPORT int64_t tpch_gen_supplier(double sf, int64_t supp_start, int64_t supp_count,
    int64_t* suppkey, uint8_t* nationkey)
{
    (void)sf;
    tpch_rng nk;
    tpch_rng_init(&nk, TPCH_SEED_S_NKEY, 1);
    tpch_rng_skip(&nk, supp_start - 1);
    for (int64_t i = 0; i < supp_count; i++) {
        if (suppkey) suppkey[i] = supp_start + i;
        int64_t v = tpch_rng_int(&nk, 0, 24);
        if (nationkey) nationkey[i] = (uint8_t)v;
        tpch_rng_row_finished(&nk);
    }
    return supp_count;
}

EXPORT int64_t tpch_gen_part_cols(double sf, int64_t part_start, int64_t part_count,
    int64_t* partkey, uint8_t* type_id)
{
    (void)sf;
    tpch_gen_part(part_start, part_count, partkey, type_id);
    return part_count;
}

/* ---- round 2: text pool + string streams (tpch_text.h) ---- */
#include "tpch_text.h"
#include <stdlib.h>

static char* g_pool = NULL;

EXPORT const char* tpch_text_pool(void)
{
    if (!g_pool) {
        g_pool = (char*)malloc(TPCH_TEXT_POOL_SIZE);
        tpch_text_pool_build(g_pool, TPCH_TEXT_POOL_SIZE);
    }
    return g_pool;
}

/* comment slices for `count` rows of a stream seeded `seed`, `per_value`
 * values per row, `usage` stream uses per row, average length `avg`.
 * offs/lens arrays sized count*per_value. */
EXPORT void tpch_text_slices(int64_t seed, int64_t row_start /*1-based*/,
                             int64_t count, int per_value, int usage, int avg,
                             int64_t* offs, int32_t* lens)
{
    tpch_rng r;
    tpch_rng_init(&r, seed, usage);
    tpch_rng_skip(&r, (row_start - 1) * usage);
    for (int64_t i = 0; i < count; i++) {
        for (int v = 0; v < per_value; v++)
            tpch_text_slice(&r, avg, &offs[i * per_value + v],
                            &lens[i * per_value + v]);
        tpch_rng_row_finished(&r);
    }
}

EXPORT void tpch_gen_vstr(int64_t seed, int64_t row_start, int64_t count,
                          int usage, int avg, char* out, int stride,
                          int32_t* lens)
{
    tpch_rng r;
    tpch_rng_init(&r, seed, usage);
    tpch_rng_skip(&r, (row_start - 1) * usage);
    for (int64_t i = 0; i < count; i++) {
        lens[i] = tpch_vstr(&r, avg, out + i * stride);
        tpch_rng_row_finished(&r);
    }
}

EXPORT void tpch_gen_part_names(int64_t part_start, int64_t count, uint8_t* ids5)
{
    tpch_rng r;
    tpch_rng_init(&r, TPCH_SEED_P_NAME, TPCH_P_NAME_USAGE);
    tpch_rng_skip(&r, (part_start - 1) * TPCH_P_NAME_USAGE);
    for (int64_t i = 0; i < count; i++) {
        tpch_part_name_ids(&r, ids5 + i * 5);
        tpch_rng_row_finished(&r);
    }
}

/* pool build with a sentence-start index (offset, draw ordinal) for the
 * table-pinning harness (tools/check_textpool.py) */
static int64_t* g_sent_off = NULL;
static int64_t* g_sent_draw = NULL;
static int64_t g_sent_n = 0;

EXPORT int64_t tpch_text_pool_sentences(int64_t** offs, int64_t** draw_ords)
{
    if (!g_sent_off) {
        int64_t cap = 16 * 1024 * 1024;
        g_sent_off = (int64_t*)malloc(cap * 8);
        g_sent_draw = (int64_t*)malloc(cap * 8);
        char* buf = (char*)malloc(TPCH_TEXT_POOL_SIZE);
        tpch_sb b = { buf, 0, TPCH_TEXT_POOL_SIZE };
        tpch_rng r;
        tpch_rng_init(&r, TPCH_SEED_TEXT, 1 << 30);
        while (b.len < TPCH_TEXT_POOL_SIZE) {
            if (g_sent_n < cap) {
                g_sent_off[g_sent_n] = b.len;
                g_sent_draw[g_sent_n] = r.used;
                g_sent_n++;
            }
            tpch_text_sentence(&b, &r);
        }
        free(buf);
    }
    *offs = g_sent_off;
    *draw_ords = g_sent_draw;
    return g_sent_n;
}
