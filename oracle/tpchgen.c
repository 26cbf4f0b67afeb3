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
    int64_t* tp_cents, int64_t* suppkey, uint8_t* shipinstruct)
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
            if (shipinstruct)  shipinstruct[n] = l.shipinstruct;
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

/* extended tables (mirror trino_amd/csrc/tpchgen.hip round-2 kernels) */
EXPORT void tpch_gen_part2(int64_t part_start, int64_t count,
    int64_t* partkey, int16_t* type_id, uint8_t* brand, int32_t* size,
    uint8_t* container, uint8_t* name_ids, int64_t* retail_cents)
{
    tpch_rng ty, mf, br, sz, cn, nm;
    tpch_rng_init(&ty, TPCH_SEED_P_TYPE, 1); tpch_rng_skip(&ty, part_start - 1);
    tpch_rng_init(&mf, TPCH_SEED_P_MFG, 1);  tpch_rng_skip(&mf, part_start - 1);
    tpch_rng_init(&br, TPCH_SEED_P_BRND, 1); tpch_rng_skip(&br, part_start - 1);
    tpch_rng_init(&sz, TPCH_SEED_P_SIZE, 1); tpch_rng_skip(&sz, part_start - 1);
    tpch_rng_init(&cn, TPCH_SEED_P_CNTR, 1); tpch_rng_skip(&cn, part_start - 1);
    tpch_rng_init(&nm, TPCH_SEED_P_NAME, TPCH_P_NAME_USAGE);
    tpch_rng_skip(&nm, (part_start - 1) * TPCH_P_NAME_USAGE);
    for (int64_t i = 0; i < count; i++) {
        int64_t p = part_start + i;
        if (partkey) partkey[i] = p;
        int64_t tid = tpch_rng_int(&ty, 1, 150) - 1;
        if (type_id) type_id[i] = (int16_t)tid;
        int64_t m = tpch_rng_int(&mf, 1, 5);
        int64_t b = m * 10 + tpch_rng_int(&br, 1, 5);
        if (brand) brand[i] = (uint8_t)b;
        if (size) size[i] = (int32_t)tpch_rng_int(&sz, 1, 50);
        if (container) container[i] = (uint8_t)(tpch_rng_int(&cn, 1, 40) - 1);
        if (name_ids) tpch_part_name_ids(&nm, name_ids + i * 5);
        tpch_rng_row_finished(&nm);
        if (retail_cents) retail_cents[i] = tpch_part_price_cents(p);
    }
}

EXPORT void tpch_gen_partsupp2(double sf, int64_t part_start, int64_t count,
    int64_t* partkey, int64_t* suppkey, int32_t* availqty, int64_t* supplycost_cents)
{
    int64_t S = (int64_t)(10000 * sf);
    tpch_rng qt, sc;
    tpch_rng_init(&qt, TPCH_SEED_PS_QTY, 4);  tpch_rng_skip(&qt, (part_start - 1) * 4);
    tpch_rng_init(&sc, TPCH_SEED_PS_SCST, 4); tpch_rng_skip(&sc, (part_start - 1) * 4);
    for (int64_t i = 0; i < count; i++) {
        int64_t p = part_start + i;
        for (int j = 0; j < 4; j++) {
            int64_t at = i * 4 + j;
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

EXPORT void tpch_gen_supplier2(double sf, int64_t supp_start, int64_t count,
    int64_t* suppkey, uint8_t* nationkey, int64_t* acctbal_cents)
{
    (void)sf;
    tpch_rng nk, ab;
    tpch_rng_init(&nk, TPCH_SEED_S_NKEY, 1); tpch_rng_skip(&nk, supp_start - 1);
    tpch_rng_init(&ab, TPCH_SEED_S_ABAL, 1); tpch_rng_skip(&ab, supp_start - 1);
    for (int64_t i = 0; i < count; i++) {
        if (suppkey) suppkey[i] = supp_start + i;
        int64_t v = tpch_rng_int(&nk, 0, 24);
        if (nationkey) nationkey[i] = (uint8_t)v;
        int64_t b = tpch_rng_int(&ab, -99999, 999999);
        if (acctbal_cents) acctbal_cents[i] = b;
        tpch_rng_row_finished(&nk);
        tpch_rng_row_finished(&ab);
    }
}

EXPORT void tpch_gen_orders3(double sf, int64_t order_start, int64_t count,
    int64_t* orderkey, int64_t* custkey, int32_t* orderdate, uint8_t* priority,
    uint8_t* orderstatus, int64_t* totalprice_cents,
    int64_t* cmnt_off, int32_t* cmnt_len, int32_t* clerk)
{
    tpch_order_streams s;
    tpch_order_streams_init(&s, sf);
    tpch_order_streams_seek(&s, order_start);
    tpch_rng cm, cl;
    tpch_rng_init(&cm, TPCH_SEED_O_CMNT, 2);
    tpch_rng_skip(&cm, (order_start - 1) * 2);
    tpch_rng_init(&cl, TPCH_SEED_O_CLRK, 1);
    tpch_rng_skip(&cl, order_start - 1);
    int64_t clerks = (int64_t)(1000 * sf);
    if (clerks < 1000) clerks = 1000;
    for (int64_t i = 0; i < count; i++) {
        tpch_order_row o;
        tpch_gen_order(&s, order_start + i, &o);
        if (orderkey)  orderkey[i] = o.orderkey;
        if (custkey)   custkey[i] = o.custkey;
        if (orderdate) orderdate[i] = o.orderdate;
        if (priority)  priority[i] = o.priority;
        if (orderstatus || totalprice_cents) {
            int64_t tp = 0;
            int nF = 0, nO = 0;
            for (int j = 0; j < o.line_count; j++) {
                tpch_lineitem_row l;
                tpch_gen_line(&s, &o, j, &l);
                tp += l.tp_cents;
                if (l.linestatus) nO++; else nF++;
            }
            if (totalprice_cents) totalprice_cents[i] = tp;
            if (orderstatus) orderstatus[i] = (nO == 0) ? 0 : (nF == 0 ? 1 : 2);
        }
        {
            int64_t off; int32_t len;
            tpch_text_slice(&cm, TPCH_CMNT_AVG_O, &off, &len);
            if (cmnt_off) cmnt_off[i] = off;
            if (cmnt_len) cmnt_len[i] = len;
        }
        if (clerk) clerk[i] = (int32_t)tpch_rng_int(&cl, 1, clerks);
        tpch_rng_row_finished(&cm);
        tpch_rng_row_finished(&cl);
        tpch_order_row_finished(&s);
    }
}

/* s_comment with BBB overlay; out strides of `stride` bytes, lens filled */
EXPORT void tpch_gen_supplier_comments(int64_t supp_start, int64_t count,
    char* out, int32_t stride, int32_t* lens)
{
    const char* pool = tpch_text_pool();
    tpch_rng cm, bs, bj, bo, bt;
    tpch_rng_init(&cm, TPCH_SEED_S_CMNT, 2);
    tpch_rng_skip(&cm, (supp_start - 1) * 2);
    tpch_rng_init(&bs, TPCH_SEED_BBB_CMNT, 1); tpch_rng_skip(&bs, supp_start - 1);
    tpch_rng_init(&bj, TPCH_SEED_BBB_JNK, 1);  tpch_rng_skip(&bj, supp_start - 1);
    tpch_rng_init(&bo, TPCH_SEED_BBB_OFF, 1);  tpch_rng_skip(&bo, supp_start - 1);
    tpch_rng_init(&bt, TPCH_SEED_BBB_TYPE, 1); tpch_rng_skip(&bt, supp_start - 1);
    for (int64_t i = 0; i < count; i++) {
        int64_t off; int32_t len;
        tpch_text_slice(&cm, TPCH_CMNT_AVG_S, &off, &len);
        char* dst = out + i * stride;
        memcpy(dst, pool + off, len);
        lens[i] = len;
        int64_t sel = tpch_rng_int(&bs, 1, 10000);
        int64_t type = tpch_rng_int(&bt, 0, 100);
        int64_t junk = tpch_rng_int(&bj, 0, len - 19);
        int64_t boff = tpch_rng_int(&bo, 0, len - (19 + junk));
        if (sel <= 10) {
            memcpy(dst + boff, "Customer ", 9);
            memcpy(dst + boff + 9 + junk,
                   (type < 50) ? "Complaints" : "Recommends", 10);
        }
        tpch_rng_row_finished(&cm);
        tpch_rng_row_finished(&bs);
        tpch_rng_row_finished(&bj);
        tpch_rng_row_finished(&bo);
        tpch_rng_row_finished(&bt);
    }
}
