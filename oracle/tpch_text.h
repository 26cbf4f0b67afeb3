/* tpch_text.h — restatement of the TPC-H dbgen text/string column streams.
 *
 * ORACLE / GENERATOR INFRASTRUCTURE, shared (like tpch_core.h) between the
 * CPU oracle and the device generator so both are bit-identical by
 * construction. The reference generates these columns via the external dep
 * io.trino.tpch:tpch v1.4 (root pom.xml:1562-1566), a faithful Java port of
 * TPC-H dbgen; this header restates the published dbgen text-generation
 * algorithm (TPC-H spec 4.2.2.10-4.2.2.13 + dists.dss text grammar).
 *
 * Every stream here is PINNED against the reference's own committed sf0.01
 * dataset (testing/trino-testing-resources/.../deltalake/.../databricks73,
 * extracted to tests/golden/tiny_sf001.json.gz — dbgen sf0.01 output,
 * verified byte-identical on the scale-independent streams to the 785/200
 * canonical SF1 rows in plugin/trino-example-http example-data) and against
 * the SF1 answer fixtures (hive_tpch q*.result). Pinned empirically in this
 * round (tests/test_tpch_text.py):
 *  - part: p_name = 5 colors, seed 709314158: fresh identity permutation,
 *    5 swaps perm[i]<->perm[UnifInt(i,91)], stream advanced to 92 uses/row;
 *    p_mfgr seed 1, p_brand mfgr*10+UnifInt(1,5) seed 46831694,
 *    p_type seed 1841581359 (6x5x5 syllables), p_size seed 1193163244
 *    UnifInt(1,50), p_container seed 727633698 over the 40-entry order below;
 *  - partsupp (per part row, 4 bridge suppliers): availqty UnifInt(1,9999)
 *    seed 1671059989 u4; supplycost cents UnifInt(100,100000) seed 1051288424
 *    u4; ps_comment avg 124 seed 1961692154 u8;
 *  - supplier: acctbal cents UnifInt(-99999,999999) seed 962338209;
 *    phone seed 884434366; address v_str avg25 seed 706178559 u9;
 *    s_comment avg 63 seed 1341315363 u2 + BBB overlay (4 aux streams);
 *  - customer: address seed 881155353 u9; c_comment avg 73 seed 1335826707;
 *  - orders: clerk UnifInt(1,max(1000,1000*SF)) seed 1171034773; o_comment
 *    avg 49 seed 276090261 u2; o_orderstatus derived from line statuses;
 *  - lineitem: shipinstruct seed 1371272478, entry order below; l_comment
 *    avg 27 seed 1095462486 u14;
 *  - alphanumeric v_str: len=UnifInt(0.4*avg,1.6*avg) then one draw per 5
 *    chars, char = ALPHA64[((seed-1) / 64^i) % 64] (alphabet pinned
 *    empirically, all 64 codes observed);
 *  - text comments: offset=UnifInt(0,POOL-maxLen), len=UnifInt(.4avg,1.6avg)
 *    into the 300 MiB grammar pool (seed 933588178, one shared stream).
 */
#ifndef TPCH_TEXT_H
#define TPCH_TEXT_H

#include "tpch_core.h"
#include <string.h>

#define TPCH_SEED_P_NAME   709314158LL
#define TPCH_SEED_P_MFG            1LL
#define TPCH_SEED_P_BRND    46831694LL
#define TPCH_SEED_P_SIZE  1193163244LL
#define TPCH_SEED_P_CNTR   727633698LL
#define TPCH_SEED_P_CMNT   804159733LL
#define TPCH_SEED_PS_QTY  1671059989LL
#define TPCH_SEED_PS_SCST 1051288424LL
#define TPCH_SEED_PS_CMNT 1961692154LL
#define TPCH_SEED_O_CLRK  1171034773LL
#define TPCH_SEED_O_CMNT   276090261LL
#define TPCH_SEED_L_CMNT  1095462486LL
#define TPCH_SEED_S_ABAL   962338209LL
#define TPCH_SEED_S_ADDR   706178559LL
#define TPCH_SEED_S_PHNE   884434366LL
#define TPCH_SEED_S_CMNT  1341315363LL
#define TPCH_SEED_C_ADDR   881155353LL
#define TPCH_SEED_C_CMNT  1335826707LL
#define TPCH_SEED_BBB_CMNT 202794285LL
#define TPCH_SEED_BBB_JNK  263032577LL
#define TPCH_SEED_BBB_OFF  715851524LL
#define TPCH_SEED_BBB_TYPE 753643799LL
#define TPCH_SEED_TEXT     933588178LL

#define TPCH_TEXT_POOL_SIZE (300LL * 1024 * 1024)

/* comment average lengths (spec text[min,max] = [0.4*avg, 1.6*avg]) */
#define TPCH_CMNT_AVG_P   14
#define TPCH_CMNT_AVG_PS 124
#define TPCH_CMNT_AVG_S   63
#define TPCH_CMNT_AVG_C   73
#define TPCH_CMNT_AVG_O   49
#define TPCH_CMNT_AVG_L   27

/* pinned: v_str alphabet indexed by ((seed-1)/64^i)%64, low digit first */
static const char TPCH_ALPHA64[65] =
    ",ZYXWVUTSRQPONMLKJIHGFEDCBA zyxwvutsrqponmlkjihgfedcba9876543210";

/* 92 colors (dists.dss `colors`, alphabetical), p_name words */
static const char* const TPCH_COLORS[92] = {
    "almond", "antique", "aquamarine", "azure", "beige", "bisque", "black",
    "blanched", "blue", "blush", "brown", "burlywood", "burnished",
    "chartreuse", "chiffon", "chocolate", "coral", "cornflower", "cornsilk",
    "cream", "cyan", "dark", "deep", "dim", "dodger", "drab", "firebrick",
    "floral", "forest", "frosted", "gainsboro", "ghost", "goldenrod",
    "green", "grey", "honeydew", "hot", "indian", "ivory", "khaki", "lace",
    "lavender", "lawn", "lemon", "light", "lime", "linen", "magenta",
    "maroon", "medium", "metallic", "midnight", "mint", "misty", "moccasin",
    "navajo", "navy", "olive", "orange", "orchid", "pale", "papaya", "peach",
    "peru", "pink", "plum", "powder", "puff", "purple", "red", "rose",
    "rosy", "royal", "saddle", "salmon", "sandy", "seashell", "sienna",
    "sky", "slate", "smoke", "snow", "spring", "steel", "tan", "thistle",
    "tomato", "turquoise", "violet", "wheat", "white", "yellow",
};

/* p_type syllables: id = (s1)*25 + (s2)*5 + s3 (0-based), pinned */
static const char* const TPCH_TYPE_S1[6] =
    { "STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO" };
static const char* const TPCH_TYPE_S2[5] =
    { "ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED" };
static const char* const TPCH_TYPE_S3[5] =
    { "TIN", "NICKEL", "BRASS", "STEEL", "COPPER" };

/* p_container entry order (pinned empirically, all 40 observed) */
static const char* const TPCH_CONTAINER_S1[5] =
    { "SM", "LG", "MED", "JUMBO", "WRAP" };
static const char* const TPCH_CONTAINER_S2[8] =
    { "CASE", "BOX", "BAG", "JAR", "PACK", "PKG", "CAN", "DRUM" };
/* container id c (1..40): s1 = (c-1)/8, s2 = (c-1)%8 */

static const char* const TPCH_SHIPINSTRUCT[4] =
    { "DELIVER IN PERSON", "COLLECT COD", "TAKE BACK RETURN", "NONE" };

static const char* const TPCH_SEGMENTS[5] =
    { "AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY", "HOUSEHOLD" };

static const char* const TPCH_PRIORITIES[5] =
    { "1-URGENT", "2-HIGH", "3-MEDIUM", "4-NOT SPECIFIED", "5-LOW" };

/* nation / region fixed tables (public 25/5-row tables; regionkey per spec) */
static const char* const TPCH_NATIONS[25] = {
    "ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
    "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ", "JAPAN",
    "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU", "CHINA", "ROMANIA",
    "SAUDI ARABIA", "VIETNAM", "RUSSIA", "UNITED KINGDOM", "UNITED STATES",
};
static const int TPCH_NATION_REGION[25] =
    { 0, 1, 1, 1, 4, 0, 3, 3, 2, 2, 4, 4, 2, 4, 0, 0, 0, 1, 2, 3,
      4, 2, 3, 3, 1 };
static const char* const TPCH_REGIONS[5] =
    { "AFRICA", "AMERICA", "ASIA", "EUROPE", "MIDDLE EAST" };

/* ---- text grammar distributions (dists.dss text section) ---- */

typedef struct { const char* s; int w; } tpch_dist_ent;

static const tpch_dist_ent TPCH_D_NOUNS[] = {
    {"packages", 40}, {"requests", 40}, {"accounts", 40}, {"deposits", 40},
    {"foxes", 20}, {"ideas", 20}, {"theodolites", 20}, {"pinto beans", 20},
    {"instructions", 20}, {"dependencies", 10}, {"excuses", 10},
    {"platelets", 10}, {"asymptotes", 10}, {"courts", 5}, {"dolphins", 5},
    {"multipliers", 1}, {"sauternes", 1}, {"warthogs", 1}, {"frets", 1},
    {"dinos", 1}, {"attainments", 1}, {"somas", 1}, {"Tiresias", 1},
    {"patterns", 1}, {"forges", 1}, {"braids", 1}, {"frays", 1},
    {"warhorses", 1}, {"dugouts", 1}, {"notornis", 1},
    {"epitaphs", 1}, {"pearls", 1}, {"tithes", 1}, {"waters", 1},
    {"orbits", 1}, {"gifts", 1}, {"sheaves", 1}, {"depths", 1},
    {"sentiments", 1}, {"decoys", 1}, {"realms", 1}, {"pains", 1},
    {"grouches", 1}, {"escapades", 1}, {"hockey players", 1},
};
static const tpch_dist_ent TPCH_D_VERBS[] = {
    {"sleep", 20}, {"wake", 20}, {"are", 20}, {"cajole", 20}, {"haggle", 20},
    {"nag", 10}, {"use", 10}, {"boost", 10}, {"affix", 5}, {"detect", 5},
    {"integrate", 5}, {"maintain", 1}, {"nod", 1}, {"was", 1}, {"lose", 1},
    {"sublate", 1}, {"solve", 1}, {"thrash", 1}, {"promise", 1},
    {"engage", 1}, {"hinder", 1}, {"print", 1}, {"x-ray", 1}, {"breach", 1},
    {"eat", 1}, {"grow", 1}, {"impress", 1}, {"mold", 1}, {"poach", 1},
    {"serve", 1}, {"run", 1}, {"dazzle", 1}, {"snooze", 1}, {"doze", 1},
    {"unwind", 1}, {"kindle", 1}, {"play", 1}, {"hang", 1}, {"believe", 1},
    {"doubt", 1},
};
/* entry order + weights solved exactly from 1004 draw-aligned constraints
 * (T=289; tools/check_textpool.py harness) */
static const tpch_dist_ent TPCH_D_ADJECTIVES[] = {
    {"special", 20}, {"pending", 20}, {"unusual", 20}, {"express", 20},
    {"furious", 1}, {"sly", 1}, {"careful", 1}, {"blithe", 1}, {"quick", 1},
    {"fluffy", 1}, {"slow", 1}, {"quiet", 1}, {"ruthless", 1}, {"thin", 1},
    {"close", 1}, {"dogged", 1}, {"daring", 1}, {"brave", 1},
    {"stealthy", 1}, {"permanent", 1}, {"enticing", 1}, {"idle", 1},
    {"busy", 1}, {"regular", 50}, {"final", 40}, {"ironic", 40},
    {"even", 30}, {"bold", 20}, {"silent", 10},
};
static const tpch_dist_ent TPCH_D_ADVERBS[] = {
    {"sometimes", 1}, {"always", 1}, {"never", 1}, {"furiously", 50},
    {"slyly", 50}, {"carefully", 50}, {"blithely", 40}, {"quickly", 30},
    {"fluffily", 20}, {"slowly", 1}, {"quietly", 1}, {"ruthlessly", 1},
    {"thinly", 1}, {"closely", 1}, {"doggedly", 1}, {"daringly", 1},
    {"bravely", 1}, {"stealthily", 1}, {"permanently", 1}, {"enticingly", 1},
    {"idly", 1}, {"busily", 1}, {"regularly", 1}, {"finally", 1},
    {"ironically", 1}, {"evenly", 1}, {"boldly", 1}, {"silently", 1},
};
static const tpch_dist_ent TPCH_D_PREPOSITIONS[] = {
    {"about", 50}, {"above", 50}, {"according to", 50}, {"across", 50},
    {"after", 50}, {"against", 40}, {"along", 40}, {"alongside of", 30},
    {"among", 30}, {"around", 20}, {"at", 10}, {"atop", 1}, {"before", 1},
    {"behind", 1}, {"beneath", 1}, {"beside", 1}, {"besides", 1},
    {"between", 1}, {"beyond", 1}, {"by", 1}, {"despite", 1}, {"during", 1},
    {"except", 1}, {"for", 1}, {"from", 1}, {"in place of", 1},
    {"inside", 1}, {"instead of", 1}, {"into", 1}, {"near", 1}, {"of", 1},
    {"on", 1}, {"outside", 1}, {"over", 1}, {"past", 1}, {"since", 1},
    {"through", 1}, {"throughout", 1}, {"to", 1}, {"toward", 1},
    {"under", 1}, {"until", 1}, {"up", 1}, {"upon", 1}, {"whithout", 1},
    {"with", 1}, {"within", 1},
};
static const tpch_dist_ent TPCH_D_AUXILLARIES[] = {
    {"do", 1}, {"may", 1}, {"might", 1}, {"shall", 1}, {"will", 1},
    {"would", 1}, {"can", 1}, {"could", 1}, {"should", 1}, {"ought to", 1},
    {"must", 1}, {"will have to", 1}, {"shall have to", 1},
    {"could have to", 1}, {"should have to", 1}, {"must have to", 1},
    {"need to", 1}, {"try to", 1},
};
static const tpch_dist_ent TPCH_D_TERMINATORS[] = {
    {".", 50}, {";", 1}, {":", 1}, {"?", 1}, {"!", 1}, {"--", 1},
};
static const tpch_dist_ent TPCH_D_GRAMMAR[] = {
    {"N V T", 3}, {"N V P T", 3}, {"N V N T", 3}, {"N P V N T", 1},
    {"N P V P T", 1},
};
static const tpch_dist_ent TPCH_D_NP[] = {
    {"N", 10}, {"J N", 20}, {"J, J N", 10}, {"D J N", 50},
};
static const tpch_dist_ent TPCH_D_VP[] = {
    {"V", 30}, {"X V", 1}, {"V D", 40}, {"X V D", 1},
};

#define TPCH_DIST_N(d) ((int)(sizeof(d) / sizeof((d)[0])))

/* weighted pick: UnifInt(1, total), first entry with cumulative >= v */
static inline const char* tpch_pick_str(const tpch_dist_ent* d, int n,
                                        tpch_rng* r)
{
    int total = 0;
    for (int i = 0; i < n; i++) total += d[i].w;
    int64_t v = tpch_rng_int(r, 1, total);
    int cum = 0;
    for (int i = 0; i < n; i++) {
        cum += d[i].w;
        if (v <= cum) return d[i].s;
    }
    return d[n - 1].s;
}

/* ---- text pool build (host only; 300 MiB, sequential) ---- */

typedef struct { char* buf; int64_t len; int64_t cap; } tpch_sb;

static inline void tpch_sb_app(tpch_sb* b, const char* s)
{
    int64_t l = (int64_t)strlen(s);
    if (b->len + l <= b->cap) memcpy(b->buf + b->len, s, l);
    b->len += l;
}

static inline void tpch_sb_ch(tpch_sb* b, char c)
{
    if (b->len < b->cap) b->buf[b->len] = c;
    b->len++;
}

static inline void tpch_text_np(tpch_sb* b, tpch_rng* r)
{
    const char* syntax = tpch_pick_str(TPCH_D_NP, TPCH_DIST_N(TPCH_D_NP), r);
    for (const char* p = syntax; *p; p++) {
        switch (*p) {
            case 'N': tpch_sb_app(b, tpch_pick_str(TPCH_D_NOUNS, TPCH_DIST_N(TPCH_D_NOUNS), r)); break;
            case 'J': tpch_sb_app(b, tpch_pick_str(TPCH_D_ADJECTIVES, TPCH_DIST_N(TPCH_D_ADJECTIVES), r)); break;
            case 'D': tpch_sb_app(b, tpch_pick_str(TPCH_D_ADVERBS, TPCH_DIST_N(TPCH_D_ADVERBS), r)); break;
            case ',': tpch_sb_ch(b, ','); break;
            case ' ': tpch_sb_ch(b, ' '); break;
            default: break;
        }
    }
}

static inline void tpch_text_vp(tpch_sb* b, tpch_rng* r)
{
    const char* syntax = tpch_pick_str(TPCH_D_VP, TPCH_DIST_N(TPCH_D_VP), r);
    for (const char* p = syntax; *p; p++) {
        switch (*p) {
            case 'V': tpch_sb_app(b, tpch_pick_str(TPCH_D_VERBS, TPCH_DIST_N(TPCH_D_VERBS), r)); break;
            case 'X': tpch_sb_app(b, tpch_pick_str(TPCH_D_AUXILLARIES, TPCH_DIST_N(TPCH_D_AUXILLARIES), r)); break;
            case 'D': tpch_sb_app(b, tpch_pick_str(TPCH_D_ADVERBS, TPCH_DIST_N(TPCH_D_ADVERBS), r)); break;
            case ' ': tpch_sb_ch(b, ' '); break;
            default: break;
        }
    }
}

static inline void tpch_text_sentence(tpch_sb* b, tpch_rng* r)
{
    const char* syntax =
        tpch_pick_str(TPCH_D_GRAMMAR, TPCH_DIST_N(TPCH_D_GRAMMAR), r);
    for (const char* p = syntax; *p; p++) {
        switch (*p) {
            case 'N': tpch_text_np(b, r); break;
            case 'V': tpch_text_vp(b, r); break;
            case 'P':
                tpch_sb_app(b, tpch_pick_str(TPCH_D_PREPOSITIONS, TPCH_DIST_N(TPCH_D_PREPOSITIONS), r));
                tpch_sb_app(b, " the ");
                tpch_text_np(b, r);
                break;
            case 'T':
                b->len--;   /* erase the space after the previous element */
                tpch_sb_app(b, tpch_pick_str(TPCH_D_TERMINATORS, TPCH_DIST_N(TPCH_D_TERMINATORS), r));
                break;
            default: continue;   /* the syntax's own spaces emit nothing */
        }
        tpch_sb_ch(b, ' ');
    }
}

/* build the full pool into buf[size]; returns bytes of last partial sentence
 * beyond size (informational) */
static inline void tpch_text_pool_build(char* buf, int64_t size)
{
    tpch_sb b = { buf, 0, size };
    tpch_rng r;
    tpch_rng_init(&r, TPCH_SEED_TEXT, 1 << 30);
    while (b.len < size) tpch_text_sentence(&b, &r);
}

/* ---- per-row value generators over independent streams ---- */

/* comment slice: 2 draws; off in [0, pool - maxLen], len in [.4avg,1.6avg] */
TPCH_HD static inline void tpch_text_slice(tpch_rng* r, int avg,
                                   int64_t* off, int32_t* len)
{
    int lo = (int)(avg * 0.4), hi = (int)(avg * 1.6);
    *off = tpch_rng_int(r, 0, TPCH_TEXT_POOL_SIZE - hi);
    *len = (int32_t)tpch_rng_int(r, lo, hi);
}

/* v_str (addresses): len then one draw per 5 chars; pinned decode */
TPCH_HD static inline int tpch_vstr(tpch_rng* r, int avg, char* out /* >= 1.6*avg */)
{
    int lo = (int)(avg * 0.4), hi = (int)(avg * 1.6);
    int len = (int)tpch_rng_int(r, lo, hi);
    int64_t v = 0;
    for (int i = 0; i < len; i++) {
        if (i % 5 == 0) v = tpch_rng_raw(r) - 1;
        out[i] = TPCH_ALPHA64[v % 64];
        v /= 64;
    }
    return len;
}

/* phone: 3 draws, "CC-LLL-LLL-LLLL" with CC = nationkey + 10 */
TPCH_HD static inline void tpch_phone(tpch_rng* r, int nationkey, char out[16])
{
    int c = 10 + nationkey;
    int l1 = (int)tpch_rng_int(r, 100, 999);
    int l2 = (int)tpch_rng_int(r, 100, 999);
    int l3 = (int)tpch_rng_int(r, 1000, 9999);
    out[0] = (char)('0' + c / 10); out[1] = (char)('0' + c % 10); out[2] = '-';
    out[3] = (char)('0' + l1 / 100); out[4] = (char)('0' + (l1 / 10) % 10);
    out[5] = (char)('0' + l1 % 10); out[6] = '-';
    out[7] = (char)('0' + l2 / 100); out[8] = (char)('0' + (l2 / 10) % 10);
    out[9] = (char)('0' + l2 % 10); out[10] = '-';
    out[11] = (char)('0' + l3 / 1000); out[12] = (char)('0' + (l3 / 100) % 10);
    out[13] = (char)('0' + (l3 / 10) % 10); out[14] = (char)('0' + l3 % 10);
    out[15] = '\0';
}

/* p_name color ids: 5 swaps of a fresh identity permutation, j=UnifInt(i,91);
 * stream advances to 92 uses per row (pinned) */
TPCH_HD static inline void tpch_part_name_ids(tpch_rng* r, uint8_t ids[5])
{
    uint8_t perm[92];
    for (int i = 0; i < 92; i++) perm[i] = (uint8_t)i;
    for (int i = 0; i < 5; i++) {
        int j = (int)tpch_rng_int(r, i, 91);
        uint8_t t = perm[i]; perm[i] = perm[j]; perm[j] = t;
        ids[i] = perm[i];
    }
}
#define TPCH_P_NAME_USAGE 92

#endif /* TPCH_TEXT_H */
