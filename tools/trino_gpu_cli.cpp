/* trino_gpu_cli — native C++ host driver over the C ABI (include/trino_gpu.h).
 *
 * The host side of this engine is C++ (the operator state machines live in
 * libtrino_gpu); this binary is the standalone driver proving the path needs
 * no Python: it chains the same operator calls LocalExecutionPlanner-built
 * factories would make for the covered TPC-H plans.
 *
 *   trino_gpu_cli q1 [sf]      fused Q1 (scan/filter/agg), prints the 4 rows
 *   trino_gpu_cli q3 [sf]      3-way join pipeline + fused dynamic filter
 *   trino_gpu_cli q6 [sf]      scan/filter/scalar agg
 *   trino_gpu_cli q13 [sf]     pool LIKE + dense-range aggregation (round 2)
 * (q4/q12/q14/q18 as well — see main.) Requires a HIP device (no CPU
 * fallback by design).
 */
#include "../include/trino_gpu.h"
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <algorithm>
#include <map>
#include <string>
#include <vector>

static void die(tg_status st, const char* what)
{
    if (st != TG_OK) {
        fprintf(stderr, "%s failed (%d): %s\n", what, st, tg_last_error());
        exit(1);
    }
}

static const char* RF = "ANR";
static const char* LS = "FO";

static int run_q1(tg_session* s, double sf)
{
    tg_tpch_lineitem_cols cols;
    die(tg_tpch_lineitem_alloc(s, sf, 1, (int64_t)(1500000 * sf), 0, &cols), "gen");
    tg_q1_result r;
    die(tg_q1_run(s, &cols, 10471, &r), "q1");
    printf("l_returnflag|l_linestatus|sum_qty|sum_base_price|sum_disc_price|sum_charge|avg_qty|avg_price|avg_disc|count_order\n");
    for (int c = 0; c < 6; c++) {
        if (!r.count[c]) continue;
        printf("%c|%c|%.2f|%.2f|%.4f|%.6f|%.6f|%.6f|%.8f|%lld\n",
               RF[c / 2], LS[c % 2], r.sum_qty[c], r.sum_base[c], r.sum_disc_price[c],
               r.sum_charge[c], r.avg_qty[c], r.avg_price[c], r.avg_disc[c],
               (long long)r.count[c]);
    }
    fprintf(stderr, "[%lld rows scanned, kernel %.3f ms, %.1f Grow/s, %.0f GB/s]\n",
            (long long)cols.row_count, r.elapsed_ms,
            cols.row_count / r.elapsed_ms / 1e6,
            cols.row_count * 38.0 / (r.elapsed_ms / 1e3) / 1e9);
    die(tg_tpch_lineitem_free(s, &cols), "free");
    return 0;
}

static tg_expr_inst I_col(int c) { tg_expr_inst i{}; i.op = TG_EXPR_COL; i.arg0 = c; return i; }
static tg_expr_inst I_i64(int64_t v) { tg_expr_inst i{}; i.op = TG_EXPR_CONST_I64; i.imm.i64 = v; return i; }
static tg_expr_inst I_f64(double v) { tg_expr_inst i{}; i.op = TG_EXPR_CONST_F64; i.imm.f64 = v; return i; }
static tg_expr_inst I_op(tg_expr_op o) { tg_expr_inst i{}; i.op = o; return i; }

static tg_page dev_page(std::vector<tg_block>& blocks, int64_t n)
{
    tg_page p{};
    p.channel_count = (int32_t)blocks.size();
    p.position_count = n;
    p.blocks = blocks.data();
    return p;
}

static tg_block dev_block(tg_type t, const void* ptr, int64_t n)
{
    tg_block b{};
    b.type = t;
    b.kind = TG_BK_VALUE;
    b.position_count = n;
    b.on_device = 1;
    b.data = ptr;
    return b;
}

static int run_q6(tg_session* s, double sf)
{
    tg_tpch_lineitem_cols li;
    die(tg_tpch_lineitem_alloc(s, sf, 1, (int64_t)(1500000 * sf), 0, &li), "gen");
    std::vector<tg_block> blocks = {
        dev_block(TG_INTEGER, li.shipdate, li.row_count),
        dev_block(TG_DOUBLE, li.quantity, li.row_count),
        dev_block(TG_DOUBLE, li.extendedprice, li.row_count),
        dev_block(TG_DOUBLE, li.discount, li.row_count),
    };
    tg_page page = dev_page(blocks, li.row_count);
    std::vector<tg_expr_inst> f = {
        I_col(0), I_i64(8766), I_op(TG_EXPR_GE),
        I_col(0), I_i64(9131), I_op(TG_EXPR_LT), I_op(TG_EXPR_AND),
        I_col(3), I_f64(0.05), I_f64(0.07), I_op(TG_EXPR_BETWEEN), I_op(TG_EXPR_AND),
        I_col(1), I_i64(24), I_op(TG_EXPR_LT), I_op(TG_EXPR_AND)};
    std::vector<tg_expr_inst> proj = {I_col(2), I_col(3), I_op(TG_EXPR_MUL)};
    tg_expr fe{f.data(), (int32_t)f.size()};
    tg_expr pe{proj.data(), (int32_t)proj.size()};
    int32_t ot = TG_DOUBLE;
    tg_operator* fp = nullptr;
    die(tg_filter_project_create(s, &fe, &pe, &ot, 1, &fp), "fp create");
    die(tg_operator_add_input(fp, &page), "fp add");
    die(tg_operator_finish(fp), "fp finish");
    tg_page sel{};
    int fin = 0;
    die(tg_operator_get_output(fp, &sel, &fin), "fp out");

    tg_agg_spec aggs[2] = {{TG_AGG_SUM_F64, 0}, {TG_AGG_COUNT_STAR, -1}};
    tg_operator* agg = nullptr;
    die(tg_hash_aggregation_create(s, nullptr, 0, nullptr, aggs, 2, TG_STEP_SINGLE, &agg), "agg create");
    die(tg_operator_add_input(agg, &sel), "agg add");
    die(tg_operator_finish(agg), "agg finish");
    tg_page out{};
    die(tg_operator_get_output(agg, &out, &fin), "agg out");
    double revenue = 0;
    int64_t rows = 0;
    die(tg_copy_dtoh(s, &revenue, out.blocks[0].data, 8), "dtoh");
    die(tg_copy_dtoh(s, &rows, out.blocks[1].data, 8), "dtoh");
    printf("revenue\n%.4f\n", revenue);
    fprintf(stderr, "[%lld rows matched]\n", (long long)rows);
    tg_operator_close(agg);
    tg_operator_close(fp);
    die(tg_tpch_lineitem_free(s, &li), "free");
    return 0;
}

static int run_q3(tg_session* s, double sf)
{
    int64_t n_orders = (int64_t)(1500000 * sf), n_cust = (int64_t)(150000 * sf);
    /* inputs */
    void *c_ck, *c_ms, *o_ok, *o_ck, *o_od;
    die(tg_device_malloc(s, &c_ck, n_cust * 8), "malloc");
    die(tg_device_malloc(s, &c_ms, n_cust), "malloc");
    die(tg_device_malloc(s, &o_ok, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_ck, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_od, n_orders * 4), "malloc");
    die(tg_tpch_gen_customer(s, sf, 1, n_cust, (int64_t*)c_ck, (uint8_t*)c_ms,
                             nullptr, nullptr), "gen cust");
    die(tg_tpch_gen_orders(s, sf, 1, n_orders, (int64_t*)o_ok, (int64_t*)o_ck, (int32_t*)o_od, nullptr), "gen ord");
    tg_tpch_lineitem_cols li;
    die(tg_tpch_lineitem_alloc(s, sf, 1, n_orders, 1, &li), "gen li");

    int fin = 0;
    /* customer WHERE mktsegment='BUILDING'(id 1) -> build1(custkey) */
    std::vector<tg_block> cb = {dev_block(TG_BIGINT, c_ck, n_cust),
                                dev_block(TG_TINYINT, c_ms, n_cust)};
    tg_page cpage = dev_page(cb, n_cust);
    std::vector<tg_expr_inst> cf = {I_col(1), I_i64(1), I_op(TG_EXPR_EQ)};
    std::vector<tg_expr_inst> cp = {I_col(0)};
    tg_expr cfe{cf.data(), (int32_t)cf.size()};
    tg_expr cpe{cp.data(), (int32_t)cp.size()};
    int32_t cot = TG_BIGINT;
    tg_operator* f1 = nullptr;
    die(tg_filter_project_create(s, &cfe, &cpe, &cot, 1, &f1), "f1");
    die(tg_operator_add_input(f1, &cpage), "f1 add");
    die(tg_operator_finish(f1), "f1 fin");
    tg_page cust_sel{};
    die(tg_operator_get_output(f1, &cust_sel, &fin), "f1 out");

    tg_join_bridge* br1 = nullptr;
    die(tg_join_bridge_create(s, &br1), "br1");
    int32_t bt1 = TG_BIGINT, kc0 = 0;
    tg_operator* b1 = nullptr;
    die(tg_hash_builder_create(s, br1, &bt1, 1, &kc0, 1, nullptr, 0, &b1), "b1");
    die(tg_operator_add_input(b1, &cust_sel), "b1 add");
    die(tg_operator_finish(b1), "b1 fin");

    /* orders WHERE orderdate < 1995-03-15 -> join customers -> build2 */
    std::vector<tg_block> ob = {dev_block(TG_BIGINT, o_ok, n_orders),
                                dev_block(TG_BIGINT, o_ck, n_orders),
                                dev_block(TG_INTEGER, o_od, n_orders)};
    tg_page opage = dev_page(ob, n_orders);
    std::vector<tg_expr_inst> of = {I_col(2), I_i64(9204), I_op(TG_EXPR_LT)};
    std::vector<tg_expr_inst> op0 = {I_col(0)}, op1 = {I_col(1)}, op2 = {I_col(2)};
    tg_expr ofe{of.data(), (int32_t)of.size()};
    tg_expr opr[3] = {{op0.data(), 1}, {op1.data(), 1}, {op2.data(), 1}};
    int32_t oot[3] = {TG_BIGINT, TG_BIGINT, TG_INTEGER};
    tg_operator* f2 = nullptr;
    die(tg_filter_project_create(s, &ofe, opr, oot, 3, &f2), "f2");
    die(tg_operator_add_input(f2, &opage), "f2 add");
    die(tg_operator_finish(f2), "f2 fin");
    tg_page ord_sel{};
    die(tg_operator_get_output(f2, &ord_sel, &fin), "f2 out");

    int32_t ptypes[3] = {TG_BIGINT, TG_BIGINT, TG_INTEGER};
    int32_t j1key = 1, j1out[2] = {0, 2};
    tg_operator* j1 = nullptr;
    die(tg_lookup_join_create(s, br1, ptypes, 3, &j1key, 1, j1out, 2, &j1), "j1");
    die(tg_operator_add_input(j1, &ord_sel), "j1 add");
    die(tg_operator_finish(j1), "j1 fin");
    tg_page ob2{};
    die(tg_operator_get_output(j1, &ob2, &fin), "j1 out");

    tg_join_bridge* br2 = nullptr;
    die(tg_join_bridge_create(s, &br2), "br2");
    /* dynamic filter: ask the build for a key-membership bitmap so the
     * lineitem scan below can drop non-matching orderkeys in-kernel
     * (DynamicPageFilter analog; round-2 ABI) */
    die(tg_join_bridge_request_bitmap(br2), "br2 df");
    int32_t bt2[2] = {TG_BIGINT, TG_INTEGER}, b2out = 1;
    tg_operator* b2 = nullptr;
    die(tg_hash_builder_create(s, br2, bt2, 2, &kc0, 1, &b2out, 1, &b2), "b2");
    die(tg_operator_add_input(b2, &ob2), "b2 add");
    die(tg_operator_finish(b2), "b2 fin");

    /* lineitem WHERE shipdate > 1995-03-15, project (okey, discprice) */
    std::vector<tg_block> lb = {dev_block(TG_BIGINT, li.orderkey, li.row_count),
                                dev_block(TG_INTEGER, li.shipdate, li.row_count),
                                dev_block(TG_DOUBLE, li.extendedprice, li.row_count),
                                dev_block(TG_DOUBLE, li.discount, li.row_count)};
    tg_page lpage = dev_page(lb, li.row_count);
    std::vector<tg_expr_inst> lf = {I_col(1), I_i64(9204), I_op(TG_EXPR_GT)};
    std::vector<tg_expr_inst> lp0 = {I_col(0)};
    std::vector<tg_expr_inst> lp1 = {I_col(2), I_f64(1.0), I_col(3), I_op(TG_EXPR_SUB),
                                     I_op(TG_EXPR_MUL)};
    tg_expr lfe{lf.data(), (int32_t)lf.size()};
    tg_expr lpr[2] = {{lp0.data(), 1}, {lp1.data(), (int32_t)lp1.size()}};
    int32_t lot[2] = {TG_BIGINT, TG_DOUBLE};
    tg_operator* f3 = nullptr;
    die(tg_filter_project_create_df(s, &lfe, lpr, lot, 2, br2, 0, &f3), "f3");
    die(tg_operator_add_input(f3, &lpage), "f3 add");
    die(tg_operator_finish(f3), "f3 fin");
    tg_page li_sel{};
    die(tg_operator_get_output(f3, &li_sel, &fin), "f3 out");

    int32_t p2types[2] = {TG_BIGINT, TG_DOUBLE}, j2out[2] = {0, 1};
    tg_operator* j2 = nullptr;
    die(tg_lookup_join_create(s, br2, p2types, 2, &kc0, 1, j2out, 2, &j2), "j2");
    die(tg_operator_add_input(j2, &li_sel), "j2 add");
    die(tg_operator_finish(j2), "j2 fin");
    tg_page joined{};
    die(tg_operator_get_output(j2, &joined, &fin), "j2 out");

    /* GROUP BY (orderkey, orderdate) SUM(discprice) -> TopN 10 */
    int32_t gch[2] = {0, 2}, gty[2] = {TG_BIGINT, TG_INTEGER};
    tg_agg_spec ag{TG_AGG_SUM_F64, 1};
    tg_operator* agg = nullptr;
    die(tg_hash_aggregation_create(s, gch, 2, gty, &ag, 1, TG_STEP_SINGLE, &agg), "agg");
    die(tg_operator_add_input(agg, &joined), "agg add");
    die(tg_operator_finish(agg), "agg fin");
    tg_page groups{};
    die(tg_operator_get_output(agg, &groups, &fin), "agg out");

    int32_t tty[3] = {TG_BIGINT, TG_INTEGER, TG_DOUBLE};
    int32_t sch[2] = {2, 1}, sdsc[2] = {1, 0};
    tg_operator* top = nullptr;
    die(tg_topn_create(s, tty, 3, sch, sdsc, 2, 10, &top), "topn");
    die(tg_operator_add_input(top, &groups), "topn add");
    die(tg_operator_finish(top), "topn fin");
    tg_page t10{};
    die(tg_operator_get_output(top, &t10, &fin), "topn out");

    printf("l_orderkey|revenue|o_orderdate|o_shippriority\n");
    for (int64_t i = 0; i < t10.position_count; i++) {
        int64_t ok;
        int32_t od;
        double rev;
        die(tg_copy_dtoh(s, &ok, (const char*)t10.blocks[0].data + i * 8, 8), "dtoh");
        die(tg_copy_dtoh(s, &od, (const char*)t10.blocks[1].data + i * 4, 4), "dtoh");
        die(tg_copy_dtoh(s, &rev, (const char*)t10.blocks[2].data + i * 8, 8), "dtoh");
        printf("%lld|%.4f|%d|0\n", (long long)ok, rev, od);
    }
    for (tg_operator* op : {f1, b1, f2, j1, b2, f3, j2, agg, top}) tg_operator_close(op);
    tg_join_bridge_close(br1);
    tg_join_bridge_close(br2);
    die(tg_tpch_lineitem_free(s, &li), "free");
    for (void* p : {c_ck, c_ms, o_ok, o_ck, o_od}) die(tg_device_free(s, p), "free");
    return 0;
}


static int run_q4(tg_session* s, double sf)
{
    /* Q4: EXISTS semi join + grouped count (python mirror:
     * trino_amd/tpch_queries.py q4_gpu) */
    int64_t n_orders = (int64_t)(1500000 * sf);
    void *o_ok, *o_od, *o_pri;
    die(tg_device_malloc(s, &o_ok, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_od, n_orders * 4), "malloc");
    die(tg_device_malloc(s, &o_pri, n_orders), "malloc");
    die(tg_tpch_gen_orders(s, sf, 1, n_orders, (int64_t*)o_ok, nullptr,
                           (int32_t*)o_od, (uint8_t*)o_pri), "gen ord");
    tg_tpch_lineitem_cols li;
    die(tg_tpch_lineitem_alloc(s, sf, 1, n_orders, 1 | 2, &li), "gen li");
    int fin = 0;

    std::vector<tg_block> lb = {dev_block(TG_BIGINT, li.orderkey, li.row_count),
                                dev_block(TG_INTEGER, li.commitdate, li.row_count),
                                dev_block(TG_INTEGER, li.receiptdate, li.row_count)};
    tg_page lpage = dev_page(lb, li.row_count);
    std::vector<tg_expr_inst> lf = {I_col(1), I_col(2), I_op(TG_EXPR_LT)};
    std::vector<tg_expr_inst> lp = {I_col(0)};
    tg_expr lfe{lf.data(), (int32_t)lf.size()};
    tg_expr lpe{lp.data(), 1};
    int32_t lot = TG_BIGINT;
    tg_operator* f1 = nullptr;
    die(tg_filter_project_create(s, &lfe, &lpe, &lot, 1, &f1), "f1");
    die(tg_operator_add_input(f1, &lpage), "f1 add");
    die(tg_operator_finish(f1), "f1 fin");
    tg_page late{};
    die(tg_operator_get_output(f1, &late, &fin), "f1 out");

    tg_join_bridge* br = nullptr;
    die(tg_join_bridge_create(s, &br), "br");
    int32_t bt = TG_BIGINT, kc0 = 0;
    tg_operator* b = nullptr;
    die(tg_hash_builder_create(s, br, &bt, 1, &kc0, 1, nullptr, 0, &b), "b");
    die(tg_operator_add_input(b, &late), "b add");
    die(tg_operator_finish(b), "b fin");

    std::vector<tg_block> ob = {dev_block(TG_BIGINT, o_ok, n_orders),
                                dev_block(TG_INTEGER, o_od, n_orders),
                                dev_block(TG_TINYINT, o_pri, n_orders)};
    tg_page opage = dev_page(ob, n_orders);
    std::vector<tg_expr_inst> of = {I_col(1), I_i64(8582), I_op(TG_EXPR_GE),
                                    I_col(1), I_i64(8674), I_op(TG_EXPR_LT),
                                    I_op(TG_EXPR_AND)};
    std::vector<tg_expr_inst> op0 = {I_col(0)}, op1 = {I_col(2)};
    tg_expr ofe{of.data(), (int32_t)of.size()};
    tg_expr opr[2] = {{op0.data(), 1}, {op1.data(), 1}};
    int32_t oot[2] = {TG_BIGINT, TG_TINYINT};
    tg_operator* f2 = nullptr;
    die(tg_filter_project_create(s, &ofe, opr, oot, 2, &f2), "f2");
    die(tg_operator_add_input(f2, &opage), "f2 add");
    die(tg_operator_finish(f2), "f2 fin");
    tg_page owin{};
    die(tg_operator_get_output(f2, &owin, &fin), "f2 out");

    tg_operator* sj = nullptr;
    die(tg_semi_join_create(s, br, 0, &sj), "sj");
    die(tg_operator_add_input(sj, &owin), "sj add");
    die(tg_operator_finish(sj), "sj fin");
    tg_page marked{};
    die(tg_operator_get_output(sj, &marked, &fin), "sj out");

    std::vector<tg_expr_inst> mf = {I_col(2), I_i64(1), I_op(TG_EXPR_EQ)};
    std::vector<tg_expr_inst> mp = {I_col(1)};
    tg_expr mfe{mf.data(), (int32_t)mf.size()};
    tg_expr mpe{mp.data(), 1};
    int32_t mot = TG_TINYINT;
    tg_operator* f3 = nullptr;
    die(tg_filter_project_create(s, &mfe, &mpe, &mot, 1, &f3), "f3");
    die(tg_operator_add_input(f3, &marked), "f3 add");
    die(tg_operator_finish(f3), "f3 fin");
    tg_page exists{};
    die(tg_operator_get_output(f3, &exists, &fin), "f3 out");

    int32_t gch = 0, gty = TG_TINYINT;
    tg_agg_spec ag{TG_AGG_COUNT_STAR, -1};
    tg_operator* agg = nullptr;
    die(tg_hash_aggregation_create(s, &gch, 1, &gty, &ag, 1, TG_STEP_SINGLE, &agg), "agg");
    die(tg_operator_add_input(agg, &exists), "agg add");
    die(tg_operator_finish(agg), "agg fin");
    tg_page out{};
    die(tg_operator_get_output(agg, &out, &fin), "agg out");

    static const char* PRI[5] = {"1-URGENT", "2-HIGH", "3-MEDIUM",
                                 "4-NOT SPECIFIED", "5-LOW"};
    struct Row { int8_t p; int64_t c; };
    std::vector<Row> rows;
    for (int64_t i = 0; i < out.position_count; i++) {
        Row r;
        die(tg_copy_dtoh(s, &r.p, (const char*)out.blocks[0].data + i, 1), "dtoh");
        die(tg_copy_dtoh(s, &r.c, (const char*)out.blocks[1].data + i * 8, 8), "dtoh");
        rows.push_back(r);
    }
    printf("o_orderpriority|order_count\n");
    for (int p = 0; p < 5; p++)
        for (auto& r : rows)
            if (r.p == p) printf("%s|%lld\n", PRI[p], (long long)r.c);
    for (tg_operator* o : {f1, b, f2, sj, f3, agg}) tg_operator_close(o);
    tg_join_bridge_close(br);
    die(tg_tpch_lineitem_free(s, &li), "free");
    for (void* pp : {o_ok, o_od, o_pri}) die(tg_device_free(s, pp), "free");
    return 0;
}

static int run_q14(tg_session* s, double sf)
{
    /* Q14: part join + conditional aggregation (python mirror: q14_gpu) */
    int64_t n_orders = (int64_t)(1500000 * sf), n_parts = (int64_t)(200000 * sf);
    void *p_pk, *p_ty;
    die(tg_device_malloc(s, &p_pk, n_parts * 8), "malloc");
    die(tg_device_malloc(s, &p_ty, n_parts * 2), "malloc");
    die(tg_tpch_gen_part(s, sf, 1, n_parts, (int64_t*)p_pk, (int16_t*)p_ty), "gen part");
    tg_tpch_lineitem_cols li;
    die(tg_tpch_lineitem_alloc(s, sf, 1, n_orders, 4, &li), "gen li");
    int fin = 0;

    std::vector<tg_block> pb = {dev_block(TG_BIGINT, p_pk, n_parts),
                                dev_block(TG_SMALLINT, p_ty, n_parts)};
    tg_page ppage = dev_page(pb, n_parts);
    tg_join_bridge* br = nullptr;
    die(tg_join_bridge_create(s, &br), "br");
    int32_t bt[2] = {TG_BIGINT, TG_SMALLINT}, kc0 = 0, bout = 1;
    tg_operator* b = nullptr;
    die(tg_hash_builder_create(s, br, bt, 2, &kc0, 1, &bout, 1, &b), "b");
    die(tg_operator_add_input(b, &ppage), "b add");
    die(tg_operator_finish(b), "b fin");

    std::vector<tg_block> lb = {dev_block(TG_BIGINT, li.partkey, li.row_count),
                                dev_block(TG_INTEGER, li.shipdate, li.row_count),
                                dev_block(TG_DOUBLE, li.extendedprice, li.row_count),
                                dev_block(TG_DOUBLE, li.discount, li.row_count)};
    tg_page lpage = dev_page(lb, li.row_count);
    std::vector<tg_expr_inst> lf = {I_col(1), I_i64(9374), I_op(TG_EXPR_GE),
                                    I_col(1), I_i64(9404), I_op(TG_EXPR_LT),
                                    I_op(TG_EXPR_AND)};
    std::vector<tg_expr_inst> lp0 = {I_col(0)};
    std::vector<tg_expr_inst> lp1 = {I_col(2), I_f64(1.0), I_col(3),
                                     I_op(TG_EXPR_SUB), I_op(TG_EXPR_MUL)};
    tg_expr lfe{lf.data(), (int32_t)lf.size()};
    tg_expr lpr[2] = {{lp0.data(), 1}, {lp1.data(), (int32_t)lp1.size()}};
    int32_t lot[2] = {TG_BIGINT, TG_DOUBLE};
    tg_operator* f = nullptr;
    die(tg_filter_project_create(s, &lfe, lpr, lot, 2, &f), "f");
    die(tg_operator_add_input(f, &lpage), "f add");
    die(tg_operator_finish(f), "f fin");
    tg_page sel{};
    die(tg_operator_get_output(f, &sel, &fin), "f out");

    int32_t ptypes[2] = {TG_BIGINT, TG_DOUBLE}, jout = 1;
    tg_operator* j = nullptr;
    die(tg_lookup_join_create(s, br, ptypes, 2, &kc0, 1, &jout, 1, &j), "j");
    die(tg_operator_add_input(j, &sel), "j add");
    die(tg_operator_finish(j), "j fin");
    tg_page joined{};
    die(tg_operator_get_output(j, &joined, &fin), "j out");

    tg_agg_spec a1{TG_AGG_SUM_F64_EXACT, 0, 43};
    tg_operator* agg1 = nullptr;
    die(tg_hash_aggregation_create(s, nullptr, 0, nullptr, &a1, 1, TG_STEP_SINGLE, &agg1), "a1");
    die(tg_operator_add_input(agg1, &joined), "a1 add");
    die(tg_operator_finish(agg1), "a1 fin");
    tg_page tot{};
    die(tg_operator_get_output(agg1, &tot, &fin), "a1 out");
    double total = 0;
    die(tg_copy_dtoh(s, &total, tot.blocks[0].data, 8), "dtoh");

    std::vector<tg_expr_inst> pf = {I_col(1), I_i64(125), I_op(TG_EXPR_GE)};
    std::vector<tg_expr_inst> pp0 = {I_col(0)};
    tg_expr pfe{pf.data(), (int32_t)pf.size()};
    tg_expr ppe{pp0.data(), 1};
    int32_t pot = TG_DOUBLE;
    tg_operator* f2 = nullptr;
    die(tg_filter_project_create(s, &pfe, &ppe, &pot, 1, &f2), "f2");
    die(tg_operator_add_input(f2, &joined), "f2 add");
    die(tg_operator_finish(f2), "f2 fin");
    tg_page promo_p{};
    die(tg_operator_get_output(f2, &promo_p, &fin), "f2 out");
    tg_agg_spec a2{TG_AGG_SUM_F64_EXACT, 0, 43};
    tg_operator* agg2 = nullptr;
    die(tg_hash_aggregation_create(s, nullptr, 0, nullptr, &a2, 1, TG_STEP_SINGLE, &agg2), "a2");
    die(tg_operator_add_input(agg2, &promo_p), "a2 add");
    die(tg_operator_finish(agg2), "a2 fin");
    tg_page pr{};
    die(tg_operator_get_output(agg2, &pr, &fin), "a2 out");
    double promo = 0;
    die(tg_copy_dtoh(s, &promo, pr.blocks[0].data, 8), "dtoh");

    printf("promo_revenue\n%.13g\n", 100.0 * promo / total);
    for (tg_operator* o : {b, f, j, agg1, f2, agg2}) tg_operator_close(o);
    tg_join_bridge_close(br);
    die(tg_tpch_lineitem_free(s, &li), "free");
    die(tg_device_free(s, p_pk), "free");
    die(tg_device_free(s, p_ty), "free");
    return 0;
}


static int run_q12(tg_session* s, double sf)
{
    /* Q12 (python mirror: q12_gpu): small filtered-lineitem build, probe
     * orders; conditional counts by shipmode. MAIL=4 SHIP=6. */
    int64_t n_orders = (int64_t)(1500000 * sf);
    void *o_ok, *o_pri;
    die(tg_device_malloc(s, &o_ok, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_pri, n_orders), "malloc");
    die(tg_tpch_gen_orders(s, sf, 1, n_orders, (int64_t*)o_ok, nullptr, nullptr,
                           (uint8_t*)o_pri), "gen ord");
    tg_tpch_lineitem_cols li;
    die(tg_tpch_lineitem_alloc(s, sf, 1, n_orders, 1 | 2 | 8, &li), "gen li");
    int fin = 0;

    std::vector<tg_block> lb = {dev_block(TG_BIGINT, li.orderkey, li.row_count),
                                dev_block(TG_TINYINT, li.shipmode, li.row_count),
                                dev_block(TG_INTEGER, li.shipdate, li.row_count),
                                dev_block(TG_INTEGER, li.commitdate, li.row_count),
                                dev_block(TG_INTEGER, li.receiptdate, li.row_count)};
    tg_page lpage = dev_page(lb, li.row_count);
    std::vector<tg_expr_inst> lf = {
        I_col(1), I_i64(4), I_op(TG_EXPR_EQ), I_col(1), I_i64(6), I_op(TG_EXPR_EQ),
        I_op(TG_EXPR_OR),
        I_col(3), I_col(4), I_op(TG_EXPR_LT), I_op(TG_EXPR_AND),
        I_col(2), I_col(3), I_op(TG_EXPR_LT), I_op(TG_EXPR_AND),
        I_col(4), I_i64(8766), I_op(TG_EXPR_GE), I_op(TG_EXPR_AND),
        I_col(4), I_i64(9131), I_op(TG_EXPR_LT), I_op(TG_EXPR_AND)};
    std::vector<tg_expr_inst> lp0 = {I_col(0)}, lp1 = {I_col(1)};
    tg_expr lfe{lf.data(), (int32_t)lf.size()};
    tg_expr lpr[2] = {{lp0.data(), 1}, {lp1.data(), 1}};
    int32_t lot[2] = {TG_BIGINT, TG_TINYINT};
    tg_operator* f = nullptr;
    die(tg_filter_project_create(s, &lfe, lpr, lot, 2, &f), "f");
    die(tg_operator_add_input(f, &lpage), "f add");
    die(tg_operator_finish(f), "f fin");
    tg_page sel{};
    die(tg_operator_get_output(f, &sel, &fin), "f out");

    tg_join_bridge* br = nullptr;
    die(tg_join_bridge_create(s, &br), "br");
    int32_t bt[2] = {TG_BIGINT, TG_TINYINT}, kc0 = 0, bout = 1;
    tg_operator* b = nullptr;
    die(tg_hash_builder_create(s, br, bt, 2, &kc0, 1, &bout, 1, &b), "b");
    die(tg_operator_add_input(b, &sel), "b add");
    die(tg_operator_finish(b), "b fin");

    std::vector<tg_block> ob = {dev_block(TG_BIGINT, o_ok, n_orders),
                                dev_block(TG_TINYINT, o_pri, n_orders)};
    tg_page opage = dev_page(ob, n_orders);
    int32_t pt[2] = {TG_BIGINT, TG_TINYINT}, jout = 1;
    tg_operator* j = nullptr;
    die(tg_lookup_join_create(s, br, pt, 2, &kc0, 1, &jout, 1, &j), "j");
    die(tg_operator_add_input(j, &opage), "j add");
    die(tg_operator_finish(j), "j fin");
    tg_page joined{};
    die(tg_operator_get_output(j, &joined, &fin), "j out");  /* (pri, mode) */

    std::vector<tg_expr_inst> pm = {I_col(1)};
    std::vector<tg_expr_inst> ph = {I_col(0), I_i64(1), I_op(TG_EXPR_LE)};
    tg_expr pr2[2] = {{pm.data(), 1}, {ph.data(), (int32_t)ph.size()}};
    int32_t pot[2] = {TG_TINYINT, TG_DOUBLE};
    tg_operator* fp = nullptr;
    die(tg_filter_project_create(s, nullptr, pr2, pot, 2, &fp), "fp");
    die(tg_operator_add_input(fp, &joined), "fp add");
    die(tg_operator_finish(fp), "fp fin");
    tg_page flagged{};
    die(tg_operator_get_output(fp, &flagged, &fin), "fp out");

    int32_t gch = 0, gty = TG_TINYINT;
    tg_agg_spec ags[2] = {{TG_AGG_SUM_F64, 1}, {TG_AGG_COUNT_STAR, -1}};
    tg_operator* agg = nullptr;
    die(tg_hash_aggregation_create(s, &gch, 1, &gty, ags, 2, TG_STEP_SINGLE, &agg), "agg");
    die(tg_operator_add_input(agg, &flagged), "agg add");
    die(tg_operator_finish(agg), "agg fin");
    tg_page out{};
    die(tg_operator_get_output(agg, &out, &fin), "agg out");
    printf("l_shipmode|high_line_count|low_line_count\n");
    struct Row { int8_t m; double high; int64_t cnt; };
    std::vector<Row> rows;
    for (int64_t i = 0; i < out.position_count; i++) {
        Row r;
        die(tg_copy_dtoh(s, &r.m, (const char*)out.blocks[0].data + i, 1), "dtoh");
        die(tg_copy_dtoh(s, &r.high, (const char*)out.blocks[1].data + i * 8, 8), "dtoh");
        die(tg_copy_dtoh(s, &r.cnt, (const char*)out.blocks[2].data + i * 8, 8), "dtoh");
        rows.push_back(r);
    }
    static const char* MODES[7] = {"REG AIR", "AIR", "RAIL", "TRUCK", "MAIL",
                                   "FOB", "SHIP"};
    for (int m = 0; m < 7; m++)
        for (auto& r : rows)
            if (r.m == m)
                printf("%s|%lld|%lld\n", MODES[m], (long long)r.high,
                       (long long)(r.cnt - (long long)r.high));
    for (tg_operator* o : {f, b, j, fp, agg}) tg_operator_close(o);
    tg_join_bridge_close(br);
    die(tg_tpch_lineitem_free(s, &li), "free");
    for (void* pp : {o_ok, o_pri}) die(tg_device_free(s, pp), "free");
    return 0;
}

static int run_q18(tg_session* s, double sf)
{
    /* Q18 (python mirror: q18_gpu): streaming aggregation over the
     * orderkey-clustered lineitem, HAVING, join orders, TopN 100. */
    int64_t n_orders = (int64_t)(1500000 * sf);
    void *o_ok, *o_ck, *o_od;
    die(tg_device_malloc(s, &o_ok, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_ck, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_od, n_orders * 4), "malloc");
    die(tg_tpch_gen_orders(s, sf, 1, n_orders, (int64_t*)o_ok, (int64_t*)o_ck,
                           (int32_t*)o_od, nullptr), "gen ord");
    tg_tpch_lineitem_cols li;
    die(tg_tpch_lineitem_alloc(s, sf, 1, n_orders, 1 | 16, &li), "gen li");
    int fin = 0;

    std::vector<tg_block> lb = {dev_block(TG_BIGINT, li.orderkey, li.row_count),
                                dev_block(TG_DOUBLE, li.quantity, li.row_count),
                                dev_block(TG_BIGINT, li.tp_cents, li.row_count)};
    tg_page lpage = dev_page(lb, li.row_count);
    tg_agg_spec sags[2] = {{TG_AGG_SUM_F64_EXACT, 1, 0}, {TG_AGG_SUM_I64, 2}};
    tg_operator* sa = nullptr;
    die(tg_streaming_aggregation_create(s, 0, sags, 2, TG_STEP_SINGLE, &sa), "sa");
    die(tg_operator_add_input(sa, &lpage), "sa add");
    die(tg_operator_finish(sa), "sa fin");
    tg_page groups{};
    die(tg_operator_get_output(sa, &groups, &fin), "sa out");

    std::vector<tg_expr_inst> hf = {I_col(1), I_f64(300.0), I_op(TG_EXPR_GT)};
    std::vector<tg_expr_inst> h0 = {I_col(0)}, h1 = {I_col(1)}, h2 = {I_col(2)};
    tg_expr hfe{hf.data(), (int32_t)hf.size()};
    tg_expr hpr[3] = {{h0.data(), 1}, {h1.data(), 1}, {h2.data(), 1}};
    int32_t hot[3] = {TG_BIGINT, TG_DOUBLE, TG_BIGINT};
    tg_operator* f = nullptr;
    die(tg_filter_project_create(s, &hfe, hpr, hot, 3, &f), "having");
    die(tg_operator_add_input(f, &groups), "having add");
    die(tg_operator_finish(f), "having fin");
    tg_page big{};
    die(tg_operator_get_output(f, &big, &fin), "having out");

    tg_join_bridge* br = nullptr;
    die(tg_join_bridge_create(s, &br), "br");
    int32_t bt[3] = {TG_BIGINT, TG_DOUBLE, TG_BIGINT}, kc0 = 0, bouts[2] = {1, 2};
    tg_operator* b = nullptr;
    die(tg_hash_builder_create(s, br, bt, 3, &kc0, 1, bouts, 2, &b), "b");
    die(tg_operator_add_input(b, &big), "b add");
    die(tg_operator_finish(b), "b fin");

    std::vector<tg_block> ob = {dev_block(TG_BIGINT, o_ok, n_orders),
                                dev_block(TG_BIGINT, o_ck, n_orders),
                                dev_block(TG_INTEGER, o_od, n_orders)};
    tg_page opage = dev_page(ob, n_orders);
    int32_t pt[3] = {TG_BIGINT, TG_BIGINT, TG_INTEGER}, jouts[3] = {0, 1, 2};
    tg_operator* j = nullptr;
    die(tg_lookup_join_create(s, br, pt, 3, &kc0, 1, jouts, 3, &j), "j");
    die(tg_operator_add_input(j, &opage), "j add");
    die(tg_operator_finish(j), "j fin");
    tg_page matched{};
    die(tg_operator_get_output(j, &matched, &fin), "j out");
    /* (okey, ckey, odate, sumqty, totcents) */

    int32_t tty[5] = {TG_BIGINT, TG_BIGINT, TG_INTEGER, TG_DOUBLE, TG_BIGINT};
    int32_t sch[2] = {4, 2}, sdsc[2] = {1, 0};
    tg_operator* top = nullptr;
    die(tg_topn_create(s, tty, 5, sch, sdsc, 2, 100, &top), "topn");
    die(tg_operator_add_input(top, &matched), "topn add");
    die(tg_operator_finish(top), "topn fin");
    tg_page t100{};
    die(tg_operator_get_output(top, &t100, &fin), "topn out");

    printf("c_name|c_custkey|o_orderkey|o_orderdate|o_totalprice|sum_qty\n");
    for (int64_t i = 0; i < t100.position_count; i++) {
        int64_t ok, ck, tc;
        int32_t od;
        double sq;
        die(tg_copy_dtoh(s, &ok, (const char*)t100.blocks[0].data + i * 8, 8), "dtoh");
        die(tg_copy_dtoh(s, &ck, (const char*)t100.blocks[1].data + i * 8, 8), "dtoh");
        die(tg_copy_dtoh(s, &od, (const char*)t100.blocks[2].data + i * 4, 4), "dtoh");
        die(tg_copy_dtoh(s, &sq, (const char*)t100.blocks[3].data + i * 8, 8), "dtoh");
        die(tg_copy_dtoh(s, &tc, (const char*)t100.blocks[4].data + i * 8, 8), "dtoh");
        printf("Customer#%09lld|%lld|%lld|%d|%lld.%02lld|%lld\n",
               (long long)ck, (long long)ck, (long long)ok, od,
               (long long)(tc / 100), (long long)(tc % 100), (long long)sq);
    }
    for (tg_operator* o : {sa, f, b, j, top}) tg_operator_close(o);
    tg_join_bridge_close(br);
    die(tg_tpch_lineitem_free(s, &li), "free");
    for (void* pp : {o_ok, o_ck, o_od}) die(tg_device_free(s, pp), "free");
    return 0;
}

static int run_q13(tg_session* s, double sf)
{
    /* Q13 (python mirror: q13_gpu, ref sql/planner tests TestTpchLocalQueries
     * shape): o_comment NOT LIKE '%special%requests%' evaluated directly over
     * the device text-pool slices, per-customer order counts and the final
     * histogram both via dense-range aggregation — the round-2 ABI
     * (tg_tpch_gen_orders3 / tg_pool_like_flags / tg_dense_aggregation_create)
     * driven from pure C++. */
    int64_t n_orders = (int64_t)(1500000 * sf), n_cust = (int64_t)(150000 * sf);
    void *o_ck, *o_coff, *o_clen, *o_flag;
    die(tg_device_malloc(s, &o_ck, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_coff, n_orders * 8), "malloc");
    die(tg_device_malloc(s, &o_clen, n_orders * 4), "malloc");
    die(tg_device_malloc(s, &o_flag, n_orders), "malloc");
    die(tg_tpch_gen_orders3(s, sf, 1, n_orders, nullptr, (int64_t*)o_ck,
                            nullptr, nullptr, nullptr, nullptr,
                            (int64_t*)o_coff, (int32_t*)o_clen), "gen ord3");
    die(tg_pool_like_flags(s, (const int64_t*)o_coff, (const int32_t*)o_clen,
                           n_orders, "%special%requests%", (uint8_t*)o_flag),
        "like");

    std::vector<tg_block> ob = {dev_block(TG_BIGINT, o_ck, n_orders),
                                dev_block(TG_TINYINT, o_flag, n_orders)};
    tg_page opage = dev_page(ob, n_orders);
    std::vector<tg_expr_inst> fe = {I_col(1), I_i64(0), I_op(TG_EXPR_EQ)};
    std::vector<tg_expr_inst> p0 = {I_col(0)};
    tg_expr fex{fe.data(), (int32_t)fe.size()};
    tg_expr prj{p0.data(), (int32_t)p0.size()};
    int32_t oty = TG_BIGINT;
    tg_operator* f = nullptr;
    die(tg_filter_project_create(s, &fex, &prj, &oty, 1, &f), "filter");
    die(tg_operator_add_input(f, &opage), "f add");
    die(tg_operator_finish(f), "f fin");
    int fin = 0;
    tg_page kept{};
    die(tg_operator_get_output(f, &kept, &fin), "f out");

    tg_agg_spec cnt{TG_AGG_COUNT_STAR, -1, 0, 0, 0, 0};
    tg_operator* a1 = nullptr;
    die(tg_dense_aggregation_create(s, 0, 1, n_cust, &cnt, &a1), "dense1");
    die(tg_operator_add_input(a1, &kept), "a1 add");
    die(tg_operator_finish(a1), "a1 fin");
    tg_page percust{};
    die(tg_operator_get_output(a1, &percust, &fin), "a1 out");
    int64_t n_with = percust.position_count;

    tg_operator* a2 = nullptr;
    die(tg_dense_aggregation_create(s, 1, 1, 4096, &cnt, &a2), "dense2");
    die(tg_operator_add_input(a2, &percust), "a2 add");
    die(tg_operator_finish(a2), "a2 fin");
    tg_page hist{};
    die(tg_operator_get_output(a2, &hist, &fin), "a2 out");

    int64_t m = hist.position_count;
    std::vector<int64_t> ccnt(m + 1), cdist(m + 1);
    if (m) {
        die(tg_copy_dtoh(s, ccnt.data(), hist.blocks[0].data, m * 8), "dtoh");
        die(tg_copy_dtoh(s, cdist.data(), hist.blocks[1].data, m * 8), "dtoh");
    }
    ccnt[m] = 0;
    cdist[m] = n_cust - n_with;  /* customers with no qualifying orders */
    std::vector<int64_t> idx(m + 1);
    for (int64_t i = 0; i <= m; i++) idx[i] = i;
    std::sort(idx.begin(), idx.end(), [&](int64_t a, int64_t b) {
        if (cdist[a] != cdist[b]) return cdist[a] > cdist[b];
        return ccnt[a] > ccnt[b];
    });
    printf("c_count|custdist\n");
    for (int64_t i : idx)
        printf("%lld|%lld\n", (long long)ccnt[i], (long long)cdist[i]);
    for (tg_operator* op : {f, a1, a2}) tg_operator_close(op);
    for (void* p : {o_ck, o_coff, o_clen, o_flag}) die(tg_device_free(s, p), "free");
    return 0;
}

static int run_q16(tg_session* s, double sf)
{
    /* Q16 (python mirror: q16_gpu): part/partsupp device generators, the
     * BBB '%Customer%Complaints%' overlay via the generic VARCHAR LIKE
     * kernel, semi-join exclusion, and count(DISTINCT ps_suppkey) per
     * (brand, type, size) through the SORTED dedup (`tg_dedup_i64`) —
     * driving the newest round-2 ABI from pure C++. */
    int64_t np_ = (int64_t)(200000 * sf), ns = (int64_t)(10000 * sf);
    void *p_pk, *p_ty, *p_br, *p_sz, *ps_pk, *ps_sk;
    die(tg_device_malloc(s, &p_pk, np_ * 8), "malloc");
    die(tg_device_malloc(s, &p_ty, np_ * 2), "malloc");
    die(tg_device_malloc(s, &p_br, np_), "malloc");
    die(tg_device_malloc(s, &p_sz, np_ * 4), "malloc");
    die(tg_device_malloc(s, &ps_pk, np_ * 4 * 8), "malloc");
    die(tg_device_malloc(s, &ps_sk, np_ * 4 * 8), "malloc");
    die(tg_tpch_gen_part2(s, sf, 1, np_, (int64_t*)p_pk, (int16_t*)p_ty,
                          (uint8_t*)p_br, (int32_t*)p_sz, nullptr, nullptr,
                          nullptr), "gen part");
    die(tg_tpch_gen_partsupp(s, sf, 1, np_, (int64_t*)ps_pk, (int64_t*)ps_sk,
                             nullptr, nullptr), "gen ps");
    /* complaint suppliers (BBB overlay) */
    int32_t* d_soff = nullptr;
    uint8_t* d_sbytes = nullptr;
    die(tg_tpch_gen_supplier_comments(s, sf, 1, ns, &d_soff, &d_sbytes),
        "gen scmnt");
    void* d_cflag = nullptr;
    die(tg_device_malloc(s, &d_cflag, ns), "malloc");
    die(tg_varchar_like_flags(s, d_sbytes, d_soff, ns,
                              "%Customer%Complaints%", (uint8_t*)d_cflag),
        "like");
    std::vector<uint8_t> cf(ns);
    die(tg_copy_dtoh(s, cf.data(), d_cflag, ns), "dtoh");
    std::vector<int64_t> bad;
    for (int64_t i = 0; i < ns; i++)
        if (cf[i]) bad.push_back(i + 1);
    tg_join_bridge* brb = nullptr;
    die(tg_join_bridge_create(s, &brb), "brb");
    int32_t bt = TG_BIGINT;
    tg_operator* bb = nullptr;
    die(tg_set_builder_create(s, brb, &bt, 1, 0, &bb), "setb");
    tg_block bblk{};
    bblk.type = TG_BIGINT;
    bblk.position_count = (int64_t)bad.size();
    bblk.data = bad.data();
    tg_page bpage{};
    bpage.channel_count = 1;
    bpage.position_count = (int64_t)bad.size();
    bpage.blocks = &bblk;
    die(tg_operator_add_input(bb, &bpage), "setb add");
    die(tg_operator_finish(bb), "setb fin");

    /* part filter: size IN (...) AND brand<>45 AND NOT type BETWEEN 65..69 */
    std::vector<tg_block> pb = {dev_block(TG_BIGINT, p_pk, np_),
                                dev_block(TG_TINYINT, p_br, np_),
                                dev_block(TG_SMALLINT, p_ty, np_),
                                dev_block(TG_INTEGER, p_sz, np_)};
    tg_page ppage = dev_page(pb, np_);
    static const int sizes[8] = {49, 14, 23, 45, 19, 3, 36, 9};
    std::vector<tg_expr_inst> fe;
    for (int i = 0; i < 8; i++) {
        fe.push_back(I_col(3));
        fe.push_back(I_i64(sizes[i]));
        fe.push_back(I_op(TG_EXPR_EQ));
        if (i) fe.push_back(I_op(TG_EXPR_OR));
    }
    fe.push_back(I_col(1)); fe.push_back(I_i64(45)); fe.push_back(I_op(TG_EXPR_NE));
    fe.push_back(I_op(TG_EXPR_AND));
    fe.push_back(I_col(2)); fe.push_back(I_i64(65)); fe.push_back(I_i64(69));
    fe.push_back(I_op(TG_EXPR_BETWEEN)); fe.push_back(I_op(TG_EXPR_NOT));
    fe.push_back(I_op(TG_EXPR_AND));
    std::vector<tg_expr_inst> c0 = {I_col(0)}, c1 = {I_col(1)}, c2 = {I_col(2)},
                              c3 = {I_col(3)};
    tg_expr fex{fe.data(), (int32_t)fe.size()};
    tg_expr prj[4] = {{c0.data(), 1}, {c1.data(), 1}, {c2.data(), 1}, {c3.data(), 1}};
    int32_t pot[4] = {TG_BIGINT, TG_TINYINT, TG_SMALLINT, TG_INTEGER};
    tg_operator* fp = nullptr;
    die(tg_filter_project_create(s, &fex, prj, pot, 4, &fp), "fp");
    die(tg_operator_add_input(fp, &ppage), "fp add");
    die(tg_operator_finish(fp), "fp fin");
    int fin = 0;
    tg_page psel{};
    die(tg_operator_get_output(fp, &psel, &fin), "fp out");
    tg_join_bridge* brp = nullptr;
    die(tg_join_bridge_create(s, &brp), "brp");
    int32_t bty[4] = {TG_BIGINT, TG_TINYINT, TG_SMALLINT, TG_INTEGER};
    int32_t kc0 = 0, bouts[3] = {1, 2, 3};
    tg_operator* bp = nullptr;
    die(tg_hash_builder_create(s, brp, bty, 4, &kc0, 1, bouts, 3, &bp), "bp");
    die(tg_operator_add_input(bp, &psel), "bp add");
    die(tg_operator_finish(bp), "bp fin");

    /* partsupp minus complaint suppliers, join part attrs */
    std::vector<tg_block> psb = {dev_block(TG_BIGINT, ps_pk, np_ * 4),
                                 dev_block(TG_BIGINT, ps_sk, np_ * 4)};
    tg_page pspage = dev_page(psb, np_ * 4);
    tg_operator* sj = nullptr;
    die(tg_semi_join_create(s, brb, 1, &sj), "semi");
    die(tg_operator_add_input(sj, &pspage), "semi add");
    die(tg_operator_finish(sj), "semi fin");
    tg_page marked{};
    die(tg_operator_get_output(sj, &marked, &fin), "semi out");
    std::vector<tg_expr_inst> nb = {I_col(2), I_i64(0), I_op(TG_EXPR_EQ)};
    std::vector<tg_expr_inst> n0 = {I_col(0)}, n1 = {I_col(1)};
    tg_expr nbe{nb.data(), (int32_t)nb.size()};
    tg_expr npr[2] = {{n0.data(), 1}, {n1.data(), 1}};
    int32_t not_[2] = {TG_BIGINT, TG_BIGINT};
    tg_operator* fnb = nullptr;
    die(tg_filter_project_create(s, &nbe, npr, not_, 2, &fnb), "fnb");
    die(tg_operator_add_input(fnb, &marked), "fnb add");
    die(tg_operator_finish(fnb), "fnb fin");
    tg_page ps_ok{};
    die(tg_operator_get_output(fnb, &ps_ok, &fin), "fnb out");
    int32_t jt[2] = {TG_BIGINT, TG_BIGINT}, jouts[1] = {1};
    tg_operator* j = nullptr;
    die(tg_lookup_join_create(s, brp, jt, 2, &kc0, 1, jouts, 1, &j), "j");
    die(tg_operator_add_input(j, &ps_ok), "j add");
    die(tg_operator_finish(j), "j fin");
    tg_page joined{};
    die(tg_operator_get_output(j, &joined, &fin), "j out");

    /* pack ((brand*160+type)*64+size)<<32 | suppkey, sorted dedup, count */
    std::vector<tg_expr_inst> pk = {I_col(1), I_i64(160), I_op(TG_EXPR_MUL),
                                    I_col(2), I_op(TG_EXPR_ADD), I_i64(64),
                                    I_op(TG_EXPR_MUL), I_col(3),
                                    I_op(TG_EXPR_ADD), I_i64(1ll << 32),
                                    I_op(TG_EXPR_MUL), I_col(0),
                                    I_op(TG_EXPR_ADD)};
    tg_expr pke{pk.data(), (int32_t)pk.size()};
    int32_t pko = TG_BIGINT;
    tg_operator* fpk = nullptr;
    die(tg_filter_project_create(s, nullptr, &pke, &pko, 1, &fpk), "pack");
    die(tg_operator_add_input(fpk, &joined), "pack add");
    die(tg_operator_finish(fpk), "pack fin");
    tg_page packed{};
    die(tg_operator_get_output(fpk, &packed, &fin), "pack out");
    void* d_dedup = nullptr;
    die(tg_device_malloc(s, &d_dedup, (packed.position_count ? packed.position_count : 1) * 8),
        "malloc");
    int64_t nuniq = 0;
    die(tg_dedup_i64(s, (const int64_t*)packed.blocks[0].data,
                     packed.position_count, 52, (int64_t*)d_dedup, &nuniq),
        "dedup");
    fprintf(stderr, "[%lld packed pairs -> %lld distinct]\n",
            (long long)packed.position_count, (long long)nuniq);

    /* count suppliers per combo (packed >> 32) on host (combos ~28k) */
    std::vector<int64_t> u(nuniq);
    die(tg_copy_dtoh(s, u.data(), d_dedup, nuniq * 8), "dtoh");
    std::map<int64_t, int> cnt;
    for (int64_t v : u) cnt[v >> 32]++;
    static const char* T1[6] = {"STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO"};
    static const char* T2[5] = {"ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"};
    static const char* T3[5] = {"TIN", "NICKEL", "BRASS", "STEEL", "COPPER"};
    /* ORDER BY supplier_cnt DESC, p_brand, p_type (string), p_size —
     * the fixture's tie-break */
    struct Row { std::string brand, type; int size, n; };
    std::vector<Row> rows;
    for (auto& kv : cnt) {
        long long combo = kv.first, szv = combo % 64, ty = (combo / 64) % 160,
                  br = combo / 64 / 160;
        char bb[16], tb[48];
        snprintf(bb, sizeof bb, "Brand#%lld", br);
        snprintf(tb, sizeof tb, "%s %s %s", T1[ty / 25], T2[(ty / 5) % 5],
                 T3[ty % 5]);
        rows.push_back({bb, tb, (int)szv, kv.second});
    }
    std::sort(rows.begin(), rows.end(), [](const Row& a, const Row& b) {
        if (a.n != b.n) return a.n > b.n;
        if (a.brand != b.brand) return a.brand < b.brand;
        if (a.type != b.type) return a.type < b.type;
        return a.size < b.size;
    });
    printf("p_brand|p_type|p_size|supplier_cnt\n");
    int shown = 0;
    for (auto& r : rows) {
        printf("%s|%s|%d|%d\n", r.brand.c_str(), r.type.c_str(), r.size, r.n);
        if (++shown >= 10) break;
    }
    for (tg_operator* o : {bb, fp, bp, sj, fnb, j, fpk}) tg_operator_close(o);
    tg_join_bridge_close(brb);
    tg_join_bridge_close(brp);
    for (void* pp : {p_pk, p_ty, p_br, p_sz, ps_pk, ps_sk, d_cflag, d_dedup})
        die(tg_device_free(s, pp), "free");
    return 0;
}

int main(int argc, char** argv)
{
    if (argc < 2 || !strcmp(argv[1], "--help")) {
        printf("usage: %s q1|q3|q4|q6|q12|q13|q14|q16|q18 [scale_factor]  (version: %s)\n",
               argv[0], tg_version());
        return argc < 2 ? 1 : 0;
    }
    double sf = argc > 2 ? atof(argv[2]) : 1.0;
    tg_session* s = nullptr;
    tg_status st = tg_session_create(0, &s);
    if (st != TG_OK) {
        fprintf(stderr, "no GPU: %s\n", tg_last_error());
        return 2;
    }
    int rc = 1;
    if (!strcmp(argv[1], "q1")) rc = run_q1(s, sf);
    else if (!strcmp(argv[1], "q3")) rc = run_q3(s, sf);
    else if (!strcmp(argv[1], "q6")) rc = run_q6(s, sf);
    else if (!strcmp(argv[1], "q4")) rc = run_q4(s, sf);
    else if (!strcmp(argv[1], "q14")) rc = run_q14(s, sf);
    else if (!strcmp(argv[1], "q12")) rc = run_q12(s, sf);
    else if (!strcmp(argv[1], "q18")) rc = run_q18(s, sf);
    else if (!strcmp(argv[1], "q13")) rc = run_q13(s, sf);
    else if (!strcmp(argv[1], "q16")) rc = run_q16(s, sf);
    else fprintf(stderr, "unknown query %s\n", argv[1]);
    tg_session_close(s);
    return rc;
}
