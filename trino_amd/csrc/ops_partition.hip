/* ops_partition.hip — partitioned output: row hash -> partition -> stable
 * per-partition split.
 *
 * Mirrors:
 *  - PagePartitioner.partitionPageByRow/ByColumn (operator/output/
 *    PagePartitioner.java:134-330): per-row partition via the canonical row
 *    hash over the partition channels (InterpretedHashGenerator.java:57-110,
 *    combine CombineHashFunction.java:29-32, bigint xxmix AbstractLongType.
 *    java:121-125) reduced by HashGenerator.processRawHash (remote hash
 *    distribution, HashGenerator.java:41-46) — the function Trino uses for
 *    the inter-worker exchange this kernel feeds (RCCL all-to-all replaces
 *    the HTTP transport; no serde, device columns move raw — SURVEY.md §5).
 *  - per-partition position lists -> per-partition pages
 *    (PositionsAppenderPageBuilder): stable (row-order-preserving) split.
 *  - null-channel row replication and RLE shortcuts are not needed on the
 *    covered plans (TPC-H exchange keys are non-null group/join keys).
 *  - tg_hash_rows: the synchronous canonical-hash helper used by parity
 *    tests (also exercises InterpretedHashGenerator semantics standalone).
 */
#include "dev_hash.h"

__global__ void k_hash_rows(const KColH* cols, int n_ch, int64_t n,
                            uint64_t* __restrict__ out)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = row_hash(cols, n_ch, i);
}

__global__ void k_partition_ids(const KColH* cols, int n_ch, int64_t n,
                                int32_t nparts, int32_t* __restrict__ pids)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        pids[i] = d_partition_remote(row_hash(cols, n_ch, i), nparts);
}

/* stable multi-partition split: chunked counts -> serial scan -> scatter */
#define PCHUNK 16384

__global__ void k_part_count(const int32_t* __restrict__ pids, int64_t n,
                             int32_t nparts, int32_t* __restrict__ counts /*[nchunks][nparts]*/)
{
    int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t nchunks = (n + PCHUNK - 1) / PCHUNK;
    if (c >= nchunks) return;
    int64_t lo = c * PCHUNK, hi = min(lo + PCHUNK, n);
    for (int64_t i = lo; i < hi; i++)
        counts[c * nparts + pids[i]]++;
}

__global__ void k_part_scan(int32_t* counts, int64_t nchunks, int32_t nparts,
                            int32_t* __restrict__ part_totals)
{
    int32_t p = threadIdx.x;
    if (p >= nparts) return;
    int32_t run = 0;
    for (int64_t c = 0; c < nchunks; c++) {
        int32_t v = counts[c * nparts + p];
        counts[c * nparts + p] = run;
        run += v;
    }
    part_totals[p] = run;
}

__global__ void k_part_scatter(const int32_t* __restrict__ pids, int64_t n,
                               int32_t nparts, const int32_t* __restrict__ chunk_offsets,
                               int32_t* const* __restrict__ out_positions)
{
    /* one wave per chunk preserves row order per partition */
    int64_t c = blockIdx.x;
    int64_t lo = c * PCHUNK, hi = min(lo + PCHUNK, n);
    if (threadIdx.x >= 64) return;
    int lane = threadIdx.x;
    /* per-wave serial walk with per-partition running offsets in LDS */
    __shared__ int32_t run[256];
    for (int32_t p = lane; p < nparts; p += 64)
        run[p] = chunk_offsets[c * nparts + p];
    for (int64_t g = lo; g < hi; g += 64) {
        int64_t i = g + lane;
        int32_t p = (i < hi) ? pids[i] : -1;
        /* lanes with the same partition must write in lane order */
        for (int32_t q = 0; q < nparts; q++) {
            unsigned long long b = __ballot(p == q);
            if (b) {
                int before = __popcll(b & ((1ull << lane) - 1ull));
                if (p == q) out_positions[q][run[q] + before] = (int32_t)i;
                if (lane == 0) run[q] += __popcll(b);
            }
        }
    }
}

tg_status run_hash_rows(tg_session* s, const DevPage& page,
                        const int32_t* channels, int32_t n_channels, uint64_t* d_hashes)
{
    KColH* d_cols = nullptr;
    tg_status st = make_kcols(s, page, channels, n_channels, &d_cols);
    if (st != TG_OK) return st;
    hipLaunchKernelGGL(k_hash_rows, dim3(tg_grid_for(page.n)), dim3(TG_BLOCK),
                       0, s->stream, d_cols, n_channels, page.n, d_hashes);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_cols);
    return TG_OK;
}

extern "C" tg_status tg_hash_rows(tg_session* s, const tg_page* page,
                                  const int32_t* channels, int32_t n_channels,
                                  uint64_t* out_hashes)
{
    if (!s || !page || !channels || !out_hashes) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    DevPage in;
    tg_status st = tg_upload_page(s, page, &in);
    if (st != TG_OK) return st;
    uint64_t* d_h = nullptr;
    TG_POOL_ALLOC(s, &d_h, (in.n ? in.n : 1) * 8);
    st = run_hash_rows(s, in, channels, n_channels, d_h);
    if (st == TG_OK) {
        TG_HIP_CHECK(hipMemcpyAsync(out_hashes, d_h, in.n * 8, hipMemcpyDeviceToHost, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    }
    tg_pool_free(s, d_h);
    tg_free_page(s, &in);
    return st;
}

struct PagePartitionerOp : tg_operator {
    std::vector<tg_type> types;
    std::vector<int32_t> partition_channels;
    int32_t nparts = 0;
    /* per partition: staged pages (appended per input page) */
    std::vector<std::vector<DevPage>> per_part;

    tg_status add_input(const tg_page* page) override
    {
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        int32_t* d_pids = nullptr;
        TG_POOL_ALLOC(s, &d_pids, (in.n ? in.n : 1) * 4);
        KColH* d_cols = nullptr;
        st = make_kcols(s, in, partition_channels.data(),
                        (int)partition_channels.size(), &d_cols);
        if (st != TG_OK) return st;
        hipLaunchKernelGGL(k_partition_ids, dim3(tg_grid_for(in.n)), dim3(TG_BLOCK),
                           0, s->stream, d_cols, (int)partition_channels.size(),
                           in.n, nparts, d_pids);
        TG_HIP_CHECK(hipGetLastError());

        int64_t nchunks = (in.n + PCHUNK - 1) / PCHUNK;
        if (nchunks < 1) nchunks = 1;
        int32_t* d_counts = nullptr;
        int32_t* d_totals = nullptr;
        TG_POOL_ALLOC(s, &d_counts, nchunks * nparts * 4);
        TG_HIP_CHECK(hipMemsetAsync(d_counts, 0, nchunks * nparts * 4, s->stream));
        TG_POOL_ALLOC(s, &d_totals, nparts * 4);
        hipLaunchKernelGGL(k_part_count, dim3(tg_grid_for(nchunks)), dim3(TG_BLOCK),
                           0, s->stream, d_pids, in.n, nparts, d_counts);
        TG_HIP_CHECK(hipGetLastError());
        hipLaunchKernelGGL(k_part_scan, dim3(1), dim3(256), 0, s->stream,
                           d_counts, nchunks, nparts, d_totals);
        TG_HIP_CHECK(hipGetLastError());
        std::vector<int32_t> totals(nparts);
        TG_HIP_CHECK(hipMemcpyAsync(totals.data(), d_totals, nparts * 4,
                                    hipMemcpyDeviceToHost, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));

        /* per-partition position buffers */
        std::vector<int32_t*> pos(nparts);
        for (int p = 0; p < nparts; p++)
            TG_POOL_ALLOC(s, &pos[p], (totals[p] ? totals[p] : 1) * 4);
        int32_t** d_pos = nullptr;
        TG_POOL_ALLOC(s, &d_pos, nparts * sizeof(int32_t*));
        TG_HIP_CHECK(hipMemcpyAsync(d_pos, pos.data(), nparts * sizeof(int32_t*),
                                    hipMemcpyHostToDevice, s->stream));
        hipLaunchKernelGGL(k_part_scatter, dim3((uint32_t)nchunks), dim3(64), 0, s->stream,
                           d_pids, in.n, nparts, d_counts, d_pos);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));

        /* gather each partition's page */
        for (int p = 0; p < nparts; p++) {
            DevPage pp;
            pp.n = totals[p];
            for (size_t c = 0; c < in.blocks.size(); c++) {
                DevBlock ob;
                st = run_gather(s, in.blocks[c], pos[p], totals[p], &ob);
                if (st != TG_OK) return st;
                pp.blocks.push_back(ob);
            }
            per_part[p].emplace_back(std::move(pp));
            tg_pool_free(s, pos[p]);
        }
        tg_pool_free(s, d_pos);
        tg_pool_free(s, d_counts);
        tg_pool_free(s, d_totals);
        tg_pool_free(s, d_pids);
        tg_pool_free(s, d_cols);
        tg_free_page(s, &in);
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        /* outputs are fetched per partition via tg_page_partitioner_get_partition */
        out->channel_count = 0;
        out->position_count = 0;
        out->blocks = nullptr;
        *finished = input_finished ? 1 : 0;
        return TG_OK;
    }

    tg_status get_partition(int32_t p, tg_page* out)
    {
        if (p < 0 || p >= nparts) { TG_SET_ERR("bad partition"); return TG_ERR_INVALID_ARG; }
        if (per_part[p].empty()) {
            out->channel_count = 0; out->position_count = 0; out->blocks = nullptr;
            return TG_OK;
        }
        /* emit the oldest staged page for this partition */
        DevPage& pg = per_part[p].front();
        out_blocks_.clear();
        for (auto& b : pg.blocks) {
            tg_block tb{};
            tb.type = b.type; tb.kind = TG_BK_VALUE;
            tb.position_count = pg.n; tb.on_device = 1;
            tb.data = b.data; tb.valid = b.valid;
            out_blocks_.push_back(tb);
        }
        out->channel_count = (int32_t)pg.blocks.size();
        out->position_count = pg.n;
        out->blocks = out_blocks_.data();
        /* move to retired list so pointers stay valid until next fetch */
        retired_.push_back(std::move(pg));
        per_part[p].erase(per_part[p].begin());
        if (retired_.size() > 2) {
            tg_free_page(s, &retired_.front());
            retired_.erase(retired_.begin());
        }
        return TG_OK;
    }

    std::vector<DevPage> retired_;

    ~PagePartitionerOp() override
    {
        for (auto& v : per_part)
            for (auto& p : v) tg_free_page(s, &p);
        for (auto& p : retired_) tg_free_page(s, &p);
    }
};

extern "C" tg_status tg_page_partitioner_create(tg_session* s,
    const int32_t* types, int32_t n_channels,
    const int32_t* partition_channels, int32_t n_partition_channels,
    int32_t partition_count, tg_operator** out)
{
    if (!s || !types || !partition_channels || partition_count < 1 ||
        partition_count > 256 || n_partition_channels < 1) {
        TG_SET_ERR("invalid partitioner spec (1..256 partitions)");
        return TG_ERR_INVALID_ARG;
    }
    auto* op = new PagePartitionerOp();
    op->s = s;
    for (int i = 0; i < n_channels; i++) op->types.push_back((tg_type)types[i]);
    op->partition_channels.assign(partition_channels,
                                  partition_channels + n_partition_channels);
    op->nparts = partition_count;
    op->per_part.resize(partition_count);
    *out = op;
    return TG_OK;
}

extern "C" tg_status tg_page_partitioner_get_partition(tg_operator* op_, int32_t partition,
                                                       tg_page* out)
{
    auto* op = dynamic_cast<PagePartitionerOp*>(op_);
    if (!op || !out) { TG_SET_ERR("not a partitioner"); return TG_ERR_INVALID_ARG; }
    return op->get_partition(partition, out);
}
