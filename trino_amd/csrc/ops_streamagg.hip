/* ops_streamagg.hip — streaming aggregation over inputs grouped by the key.
 *
 * Mirrors operator/StreamingAggregationOperator.java: when the input is
 * sorted/clustered on the group key (the planner proves it; here the caller
 * asserts it — e.g. lineitem is clustered by orderkey), no hash table is
 * needed: group boundaries are run boundaries, group ids are run indexes
 * (trivially in first-occurrence row order), and runs that span page edges
 * continue the previous group (StreamingAggregationOperator.java:179-214).
 *
 * Q18's first aggregation (600M lineitem rows -> 150M orderkey groups at
 * SF100) costs ~700 ms through the hash path (random probes over a >10 GB
 * table); this path is three bandwidth-bound passes (~3 ms/GB).
 *
 * Round-1 scope: single BIGINT key without nulls (fails loudly otherwise),
 * SINGLE/PARTIAL step semantics identical (state emit = finals for the
 * supported aggs except AVG, which emits (count,sum) on PARTIAL like the
 * hash operator).
 */
#include "dev_hash.h"
#include "agg_common.h"

__global__ void k_run_flags(const int64_t* __restrict__ keys, int64_t n,
                            int64_t prev_last_key, int has_prev,
                            int32_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int64_t prev = (i == 0) ? prev_last_key : keys[i - 1];
        int is_new = (i == 0 && !has_prev) ? 1 : (keys[i] != prev ? 1 : 0);
        flags[i] = is_new;
    }
}

/* after the exclusive scan, arr[i] holds "runs before row i"; the row's run
 * id is that value MINUS (1 - flag...) — recover flag from neighbors:
 * run_id[i] = scanned[i] + flag[i] - 1 where flag = scanned[i+1]-scanned[i]
 * (or total at the end). Simpler: keep flags in a second array. */
__global__ void k_run_ids(const int32_t* __restrict__ flags_scanned,
                          const int32_t* __restrict__ flags, int64_t n,
                          int64_t run_base, int32_t* __restrict__ gids,
                          const int64_t* __restrict__ keys,
                          int64_t* __restrict__ keys_by_run)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        /* inclusive run count at i = exclusive scan + flag */
        int32_t rid = flags_scanned[i] + flags[i] - 1;   /* ids 0-based; a
            continued first run (flag 0, scan 0) yields run_base-1 == the
            previous page's last run */
        int64_t g = run_base + rid;
        gids[i] = (int32_t)g;
        if (flags[i]) keys_by_run[g] = keys[i];
    }
}

struct StreamAggOp : tg_operator {
    int32_t key_channel = 0;
    std::vector<tg_agg_spec> aggs;
    tg_agg_step step = TG_STEP_SINGLE;

    std::vector<KAgg> agg_state;
    int64_t* d_keys_by_run = nullptr;
    int64_t runs_cap = 0;
    int64_t n_runs = 0;            /* completed + open runs so far */
    int64_t rows_seen = 0;
    int64_t prev_last_key = 0;
    int has_prev = 0;
    bool emitted = false;

    tg_status ensure_runs(int64_t need)
    {
        if (need <= runs_cap) return TG_OK;
        int64_t ncap = runs_cap ? runs_cap : (1 << 16);
        while (ncap < need) ncap *= 2;
        int64_t* nk = nullptr;
        TG_POOL_ALLOC(s, &nk, ncap * 8);
        if (n_runs) {
            TG_HIP_CHECK(hipMemcpyAsync(nk, d_keys_by_run, n_runs * 8,
                                        hipMemcpyDeviceToDevice, s->stream));
        }
        for (auto& a : agg_state) {
            if (a.sum) {
                double* ns = nullptr;
                TG_POOL_ALLOC(s, &ns, ncap * 8);
                TG_HIP_CHECK(hipMemsetAsync(ns, 0, ncap * 8, s->stream));
                if (n_runs) {
                    TG_HIP_CHECK(hipMemcpyAsync(ns, a.sum, n_runs * 8,
                                                hipMemcpyDeviceToDevice, s->stream));
                }
                TG_HIP_CHECK(hipStreamSynchronize(s->stream));
                if (runs_cap) tg_pool_free(s, a.sum);
                a.sum = ns;
            }
            if (a.cnt) {
                long long* nc = nullptr;
                TG_POOL_ALLOC(s, &nc, ncap * 8);
                TG_HIP_CHECK(hipMemsetAsync(nc,
                        a.fn == TG_AGG_MIN_I64 ? 0xFF : 0, ncap * 8, s->stream));
                if (n_runs) {
                    TG_HIP_CHECK(hipMemcpyAsync(nc, a.cnt, n_runs * 8,
                                                hipMemcpyDeviceToDevice, s->stream));
                }
                TG_HIP_CHECK(hipStreamSynchronize(s->stream));
                if (runs_cap) tg_pool_free(s, a.cnt);
                a.cnt = nc;
            }
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        if (d_keys_by_run) tg_pool_free(s, d_keys_by_run);
        d_keys_by_run = nk;
        runs_cap = ncap;
        return TG_OK;
    }

    tg_status add_input(const tg_page* page) override
    {
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        if (in.n == 0) { tg_free_page(s, &in); return TG_OK; }
        const DevBlock& kb = in.blocks[key_channel];
        if (kb.type != TG_BIGINT || kb.valid) {
            tg_free_page(s, &in);
            TG_SET_ERR("streaming aggregation: single non-null BIGINT key in round 1");
            return TG_ERR_UNSUPPORTED;
        }
        const int64_t* keys = (const int64_t*)kb.data;
        int32_t* d_flags = nullptr;
        int32_t* d_scan = nullptr;
        int32_t* d_gids = nullptr;
        int32_t* d_total = nullptr;
        TG_POOL_ALLOC(s, &d_flags, in.n * 4);
        TG_POOL_ALLOC(s, &d_scan, in.n * 4);
        TG_POOL_ALLOC(s, &d_gids, in.n * 4);
        TG_POOL_ALLOC(s, &d_total, 4);
        hipLaunchKernelGGL(k_run_flags, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                           0, s->stream, keys, in.n, prev_last_key, has_prev, d_flags);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipMemcpyAsync(d_scan, d_flags, in.n * 4,
                                    hipMemcpyDeviceToDevice, s->stream));
        st = run_scan_i32(s, d_scan, in.n, d_total);
        if (st != TG_OK) return st;
        int32_t new_runs = 0;
        TG_HIP_CHECK(hipMemcpy(&new_runs, d_total, 4, hipMemcpyDeviceToHost));
        /* rid = scanned+flag-1 makes a continued first run (flag 0) resolve
         * to run_base-1 == the previous page's last run; new runs start at
         * run_base. So run_base is simply the run count so far. */
        int64_t run_base = n_runs;
        st = ensure_runs(run_base + new_runs + 1);
        if (st != TG_OK) return st;
        hipLaunchKernelGGL(k_run_ids, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                           0, s->stream, d_scan, d_flags, in.n, run_base, d_gids,
                           keys, d_keys_by_run);
        TG_HIP_CHECK(hipGetLastError());

        KColH* d_all = nullptr;
        st = make_kcols(s, in, nullptr, (int)in.blocks.size(), &d_all);
        if (st == TG_OK && !aggs.empty()) {
            KAgg* d_aggs = nullptr;
            TG_POOL_ALLOC(s, &d_aggs, agg_state.size() * sizeof(KAgg));
            TG_HIP_CHECK(hipMemcpyAsync(d_aggs, agg_state.data(),
                                        agg_state.size() * sizeof(KAgg),
                                        hipMemcpyHostToDevice, s->stream));
            hipLaunchKernelGGL(k_agg_update_sorted, dim3(tg_grid_for(in.n)),
                               dim3(TG_BLOCK), 0, s->stream, d_gids, in.n, d_all,
                               d_aggs, (int)agg_state.size(),
                               step == TG_STEP_FINAL ? 1 : 0);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            tg_pool_free(s, d_aggs);
        }
        TG_HIP_CHECK(hipMemcpy(&prev_last_key, keys + (in.n - 1), 8,
                               hipMemcpyDeviceToHost));
        has_prev = 1;
        n_runs = run_base + new_runs;   /* continuation adds no run */
        rows_seen += in.n;
        tg_pool_free(s, d_flags);
        tg_pool_free(s, d_scan);
        tg_pool_free(s, d_gids);
        tg_pool_free(s, d_total);
        if (d_all) tg_pool_free(s, d_all);
        tg_free_page(s, &in);
        return st;
    }

    tg_status emit()
    {
        int64_t ng = n_runs;
        DevPage outp;
        outp.n = ng;
        int grid = (int)((ng + TG_BLOCK - 1) / TG_BLOCK);
        if (grid < 1) grid = 1;
        /* key channel: identity order (run order == first-occurrence order) */
        DevBlock kb;
        kb.type = TG_BIGINT;
        kb.n = ng;
        TG_POOL_ALLOC(s, &kb.data, (ng ? ng : 1) * 8);
        if (ng) {
            TG_HIP_CHECK(hipMemcpyAsync(kb.data, d_keys_by_run, ng * 8,
                                        hipMemcpyDeviceToDevice, s->stream));
        }
        outp.blocks.push_back(kb);
        st_emit_aggs(&outp, ng);
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        stage_output(std::move(outp));
        return TG_OK;
    }

    void st_emit_aggs(DevPage* outp, int64_t ng);

    tg_status get_output(tg_page* out, int* finished) override
    {
        if (input_finished && !emitted) {
            emitted = true;
            tg_status st = emit();
            if (st != TG_OK) return st;
        }
        emit_staged(out, finished);
        return TG_OK;
    }

    ~StreamAggOp() override
    {
        if (d_keys_by_run) tg_pool_free(s, d_keys_by_run);
        for (auto& a : agg_state) {
            if (a.sum) tg_pool_free(s, a.sum);
            if (a.cnt) tg_pool_free(s, a.cnt);
        }
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

/* identity emits (no remap): reuse the hash operator's emit kernels with an
 * identity old_by_new */
__global__ void k_iota_i32(int32_t* v, int64_t n)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) v[i] = (int32_t)i;
}

void StreamAggOp::st_emit_aggs(DevPage* outp, int64_t ng)
{
    int32_t* d_obn = nullptr;
    (void)tg_pool_alloc(s, (void**)&d_obn, (ng ? ng : 1) * 4);
    hipLaunchKernelGGL(k_iota_i32, dim3(tg_grid_for(ng ? ng : 1)), dim3(TG_BLOCK),
                       0, s->stream, d_obn, ng);
    int grid = (int)((ng + TG_BLOCK - 1) / TG_BLOCK);
    if (grid < 1) grid = 1;
    for (size_t a = 0; a < agg_state.size(); a++) {
        const KAgg& ag = agg_state[a];
        bool final_out = (step != TG_STEP_PARTIAL);
        if (ag.fn == TG_AGG_AVG_F64 && !final_out) {
            DevBlock bc; bc.type = TG_BIGINT; bc.n = ng;
            (void)tg_pool_alloc(s, (void**)&bc.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.cnt, d_obn, (int32_t)ng, (int64_t*)bc.data);
            outp->blocks.push_back(bc);
            DevBlock bs; bs.type = TG_DOUBLE; bs.n = ng;
            (void)tg_pool_alloc(s, (void**)&bs.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_f64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.sum, d_obn, (int32_t)ng, (double*)bs.data);
            outp->blocks.push_back(bs);
        }
        else if (ag.fn == TG_AGG_AVG_F64) {
            DevBlock b; b.type = TG_DOUBLE; b.n = ng;
            (void)tg_pool_alloc(s, (void**)&b.data, (int64_t)(ng ? ng : 1) * 8);
            int64_t words = (ng + 63) / 64;
            (void)tg_pool_alloc(s, (void**)&b.valid, (words ? words : 1) * 8);
            (void)hipMemsetAsync(b.valid, 0xFF, words * 8, s->stream);
            hipLaunchKernelGGL(k_emit_avg, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.sum, ag.cnt, d_obn, (int32_t)ng, (double*)b.data,
                               b.valid);
            outp->blocks.push_back(b);
        }
        else if (ag.fn == TG_AGG_SUM_F64_EXACT && final_out) {
            DevBlock b; b.type = TG_DOUBLE; b.n = ng;
            (void)tg_pool_alloc(s, (void**)&b.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_exact, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               (const unsigned long long*)ag.cnt,
                               (const unsigned long long*)ag.sum, d_obn, (int32_t)ng,
                               1.0 / ag.scale, (double*)b.data);
            outp->blocks.push_back(b);
        }
        else if (ag.fn == TG_AGG_SUM_F64_EXACT) {
            DevBlock bl; bl.type = TG_BIGINT; bl.n = ng;
            (void)tg_pool_alloc(s, (void**)&bl.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.cnt, d_obn, (int32_t)ng, (int64_t*)bl.data);
            outp->blocks.push_back(bl);
            DevBlock bh; bh.type = TG_BIGINT; bh.n = ng;
            (void)tg_pool_alloc(s, (void**)&bh.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               (const long long*)ag.sum, d_obn, (int32_t)ng,
                               (int64_t*)bh.data);
            outp->blocks.push_back(bh);
        }
        else if (ag.fn == TG_AGG_MIN_I64 || ag.fn == TG_AGG_MAX_I64) {
            DevBlock b; b.type = TG_BIGINT; b.n = ng;
            (void)tg_pool_alloc(s, (void**)&b.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_i64_biased, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.cnt, d_obn, (int32_t)ng, (int64_t*)b.data);
            outp->blocks.push_back(b);
        }
        else if (ag.fn == TG_AGG_SUM_F64) {
            DevBlock b; b.type = TG_DOUBLE; b.n = ng;
            (void)tg_pool_alloc(s, (void**)&b.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_f64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.sum, d_obn, (int32_t)ng, (double*)b.data);
            outp->blocks.push_back(b);
        }
        else {
            DevBlock b; b.type = TG_BIGINT; b.n = ng;
            (void)tg_pool_alloc(s, (void**)&b.data, (int64_t)(ng ? ng : 1) * 8);
            hipLaunchKernelGGL(k_emit_i64, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                               ag.cnt, d_obn, (int32_t)ng, (int64_t*)b.data);
            outp->blocks.push_back(b);
        }
    }
    tg_pool_free(s, d_obn);
}

extern "C" tg_status tg_streaming_aggregation_create(tg_session* s,
    int32_t key_channel, const tg_agg_spec* aggs, int32_t n_aggs,
    int32_t step, tg_operator** out)
{
    if (!s || !out || n_aggs < 0 || n_aggs > 12) {
        TG_SET_ERR("invalid streaming aggregation spec");
        return TG_ERR_INVALID_ARG;
    }
    auto* op = new StreamAggOp();
    op->s = s;
    op->key_channel = key_channel;
    op->step = (tg_agg_step)step;
    for (int a = 0; a < n_aggs; a++) {
        op->aggs.push_back(aggs[a]);
        KAgg k{};
        k.fn = aggs[a].fn;
        k.in_ch = aggs[a].input_channel;
        k.mask_a = aggs[a].mask_gt_a;
        k.mask_b = aggs[a].mask_gt_b;
        k.scale = 1.0;
        for (int32_t sp_ = 0; sp_ < aggs[a].scale_pow; sp_++) k.scale *= 2.0;
        k.sum = nullptr;
        k.cnt = nullptr;
        op->agg_state.push_back(k);
    }
    /* state arrays are allocated lazily by ensure_runs; mark which are used */
    for (auto& k : op->agg_state) {
        bool needs_sum = (k.fn == TG_AGG_SUM_F64 || k.fn == TG_AGG_AVG_F64 ||
                          k.fn == TG_AGG_SUM_F64_EXACT);
        bool needs_cnt = (k.fn != TG_AGG_SUM_F64);
        k.sum = needs_sum ? (double*)(uintptr_t)1 : nullptr;    /* sentinel */
        k.cnt = needs_cnt ? (long long*)(uintptr_t)1 : nullptr;
    }
    /* replace sentinels with real (initial) allocations */
    tg_status st = op->ensure_runs(1 << 16);
    if (st != TG_OK) { delete op; return st; }
    *out = op;
    return TG_OK;
}
