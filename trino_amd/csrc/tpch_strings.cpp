/* tpch_strings.cpp — host-side materialization of TPC-H string columns for
 * final output assembly (the few rows a query RETURNS, not the scan path).
 * Mirrors what plugin/trino-tpch's TpchRecordSet.java surfaces for VARCHAR
 * columns; streams restated in oracle/tpch_text.h (pinned, see header).
 *
 * Product code: compiled into libtrino_gpu; does NOT link oracle object
 * code (tpch_text.h is the shared header-only restatement, like
 * tpch_core.h since round 1).
 */
#include "common.h"

#define TPCH_HD
#include "../../oracle/tpch_text.h"

#include <cstdlib>

extern const char* tg_tpch_host_pool(void);

/* fill caller buffers (stride bytes apart, NUL-terminated) for each key in
 * keys[n]. Any output pointer may be NULL. Keys are 1-based row keys. */
extern "C" tg_status tg_tpch_supplier_strings(double sf, const int64_t* keys,
    int32_t n, int32_t stride, char* name, char* address, char* phone,
    char* comment, int64_t* acctbal_cents, int32_t* nationkey)
{
    (void)sf;
    const char* pool = (comment) ? tg_tpch_host_pool() : nullptr;
    for (int32_t i = 0; i < n; i++) {
        int64_t k = keys[i];
        if (name)
            snprintf(name + (int64_t)i * stride, stride, "Supplier#%09lld",
                     (long long)k);
        if (address) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_S_ADDR, 9);
            tpch_rng_skip(&r, (k - 1) * 9);
            char buf[64];
            int len = tpch_vstr(&r, 25, buf);
            memcpy(address + (int64_t)i * stride, buf, len);
            address[(int64_t)i * stride + len] = 0;
        }
        int nk = 0;
        {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_S_NKEY, 1);
            tpch_rng_skip(&r, k - 1);
            nk = (int)tpch_rng_int(&r, 0, 24);
            if (nationkey) nationkey[i] = nk;
        }
        if (phone) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_S_PHNE, 3);
            tpch_rng_skip(&r, (k - 1) * 3);
            char buf[16];
            tpch_phone(&r, nk, buf);
            memcpy(phone + (int64_t)i * stride, buf, 16);
        }
        if (acctbal_cents) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_S_ABAL, 1);
            tpch_rng_skip(&r, k - 1);
            acctbal_cents[i] = tpch_rng_int(&r, -99999, 999999);
        }
        if (comment) {
            tpch_rng r, bs, bj, bo, bt;
            tpch_rng_init(&r, TPCH_SEED_S_CMNT, 2);
            tpch_rng_skip(&r, (k - 1) * 2);
            tpch_rng_init(&bs, TPCH_SEED_BBB_CMNT, 1); tpch_rng_skip(&bs, k - 1);
            tpch_rng_init(&bj, TPCH_SEED_BBB_JNK, 1);  tpch_rng_skip(&bj, k - 1);
            tpch_rng_init(&bo, TPCH_SEED_BBB_OFF, 1);  tpch_rng_skip(&bo, k - 1);
            tpch_rng_init(&bt, TPCH_SEED_BBB_TYPE, 1); tpch_rng_skip(&bt, k - 1);
            int64_t off; int32_t len;
            tpch_text_slice(&r, TPCH_CMNT_AVG_S, &off, &len);
            char* dst = comment + (int64_t)i * stride;
            memcpy(dst, pool + off, len);
            dst[len] = 0;
            int64_t sel = tpch_rng_int(&bs, 1, 10000);
            int64_t type = tpch_rng_int(&bt, 0, 100);
            int64_t junk = tpch_rng_int(&bj, 0, len - 19);
            int64_t boff = tpch_rng_int(&bo, 0, len - (19 + junk));
            if (sel <= 10) {
                memcpy(dst + boff, "Customer ", 9);
                memcpy(dst + boff + 9 + junk,
                       (type < 50) ? "Complaints" : "Recommends", 10);
            }
        }
    }
    return TG_OK;
}

extern "C" tg_status tg_tpch_customer_strings(double sf, const int64_t* keys,
    int32_t n, int32_t stride, char* name, char* address, char* phone,
    char* comment, int64_t* acctbal_cents, int32_t* nationkey)
{
    (void)sf;
    const char* pool = (comment) ? tg_tpch_host_pool() : nullptr;
    for (int32_t i = 0; i < n; i++) {
        int64_t k = keys[i];
        if (name)
            snprintf(name + (int64_t)i * stride, stride, "Customer#%09lld",
                     (long long)k);
        if (address) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_C_ADDR, 9);
            tpch_rng_skip(&r, (k - 1) * 9);
            char buf[64];
            int len = tpch_vstr(&r, 25, buf);
            memcpy(address + (int64_t)i * stride, buf, len);
            address[(int64_t)i * stride + len] = 0;
        }
        int nk = 0;
        {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_C_NKEY, 1);
            tpch_rng_skip(&r, k - 1);
            nk = (int)tpch_rng_int(&r, 0, 24);
            if (nationkey) nationkey[i] = nk;
        }
        if (phone) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_C_PHNE, 3);
            tpch_rng_skip(&r, (k - 1) * 3);
            char buf[16];
            tpch_phone(&r, nk, buf);
            memcpy(phone + (int64_t)i * stride, buf, 16);
        }
        if (acctbal_cents) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_C_ABAL, 1);
            tpch_rng_skip(&r, k - 1);
            acctbal_cents[i] = tpch_rng_int(&r, -99999, 999999);
        }
        if (comment) {
            tpch_rng r;
            tpch_rng_init(&r, TPCH_SEED_C_CMNT, 2);
            tpch_rng_skip(&r, (k - 1) * 2);
            int64_t off; int32_t len;
            tpch_text_slice(&r, TPCH_CMNT_AVG_C, &off, &len);
            char* dst = comment + (int64_t)i * stride;
            memcpy(dst, pool + off, len);
            dst[len] = 0;
        }
    }
    return TG_OK;
}

/* part strings: mfgr "Manufacturer#M", brand "Brand#MN", type text,
 * container text, name (5 colors) */
extern "C" tg_status tg_tpch_part_strings(double sf, const int64_t* keys,
    int32_t n, int32_t stride, char* name, char* mfgr, char* brand,
    char* type, char* container)
{
    (void)sf;
    for (int32_t i = 0; i < n; i++) {
        int64_t k = keys[i];
        tpch_rng r;
        if (mfgr || brand) {
            tpch_rng_init(&r, TPCH_SEED_P_MFG, 1);
            tpch_rng_skip(&r, k - 1);
            int m = (int)tpch_rng_int(&r, 1, 5);
            if (mfgr)
                snprintf(mfgr + (int64_t)i * stride, stride, "Manufacturer#%d", m);
            if (brand) {
                tpch_rng_init(&r, TPCH_SEED_P_BRND, 1);
                tpch_rng_skip(&r, k - 1);
                int b = m * 10 + (int)tpch_rng_int(&r, 1, 5);
                snprintf(brand + (int64_t)i * stride, stride, "Brand#%d", b);
            }
        }
        if (type) {
            tpch_rng_init(&r, TPCH_SEED_P_TYPE, 1);
            tpch_rng_skip(&r, k - 1);
            int t = (int)tpch_rng_int(&r, 1, 150) - 1;
            snprintf(type + (int64_t)i * stride, stride, "%s %s %s",
                     TPCH_TYPE_S1[t / 25], TPCH_TYPE_S2[(t / 5) % 5],
                     TPCH_TYPE_S3[t % 5]);
        }
        if (container) {
            tpch_rng_init(&r, TPCH_SEED_P_CNTR, 1);
            tpch_rng_skip(&r, k - 1);
            int c = (int)tpch_rng_int(&r, 1, 40) - 1;
            snprintf(container + (int64_t)i * stride, stride, "%s %s",
                     TPCH_CONTAINER_S1[c / 8], TPCH_CONTAINER_S2[c % 8]);
        }
        if (name) {
            tpch_rng_init(&r, TPCH_SEED_P_NAME, TPCH_P_NAME_USAGE);
            tpch_rng_skip(&r, (k - 1) * TPCH_P_NAME_USAGE);
            uint8_t ids[5];
            tpch_part_name_ids(&r, ids);
            snprintf(name + (int64_t)i * stride, stride, "%s %s %s %s %s",
                     TPCH_COLORS[ids[0]], TPCH_COLORS[ids[1]],
                     TPCH_COLORS[ids[2]], TPCH_COLORS[ids[3]],
                     TPCH_COLORS[ids[4]]);
        }
    }
    return TG_OK;
}

extern "C" tg_status tg_tpch_nation_name(int32_t nationkey, char out[32])
{
    if (nationkey < 0 || nationkey > 24) { TG_SET_ERR("bad nationkey"); return TG_ERR_INVALID_ARG; }
    snprintf(out, 32, "%s", TPCH_NATIONS[nationkey]);
    return TG_OK;
}

extern "C" int32_t tg_tpch_nation_region(int32_t nationkey)
{
    return (nationkey >= 0 && nationkey <= 24) ? TPCH_NATION_REGION[nationkey] : -1;
}
