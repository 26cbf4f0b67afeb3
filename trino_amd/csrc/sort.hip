/* sort.hip — device radix sort helpers (rocPRIM) shared by the group-by
 * output remap (HashAggOp::emit) and the TopN operator.
 *
 * Both previously sorted on the host (std::sort / std::partial_sort over
 * ~10^6 entries = tens of ms per Q3 step, dwarfing the kernels they fed).
 * rocprim::radix_sort_pairs is stable, which the multi-key TopN relies on
 * (least-significant-key-first passes == lexicographic order).
 */
#include "common.h"
#include "operators.h"
#include <rocprim/rocprim.hpp>

__global__ void k_iota_i64(int64_t* v, int64_t n)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) v[i] = i;
}

__global__ void k_i64_to_i32(const int64_t* __restrict__ in, int64_t n,
                             int32_t* __restrict__ out)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = (int32_t)in[i];
}

/* stable ascending sort of (u64 key, i64 value) pairs; keys/vals updated
 * in place (double-buffered internally). `bits` limits the radix passes
 * when the key range is known (e.g. pool offsets < 2^29). */
tg_status run_sort_pairs_bits(tg_session* s, uint64_t* d_keys, int64_t* d_vals,
                              int64_t n, int bits)
{
    if (n <= 1) return TG_OK;
    uint64_t* d_keys2 = nullptr;
    int64_t* d_vals2 = nullptr;
    TG_POOL_ALLOC(s, &d_keys2, n * 8);
    TG_POOL_ALLOC(s, &d_vals2, n * 8);
    size_t temp_bytes = 0;
    hipError_t e0 = rocprim::radix_sort_pairs(nullptr, temp_bytes, d_keys, d_keys2,
                                              d_vals, d_vals2, (size_t)n, 0, bits,
                                              s->stream);
    if (e0 != hipSuccess) { TG_SET_ERR("rocprim size query: %s", hipGetErrorName(e0)); return TG_ERR_HIP; }
    void* d_temp = nullptr;
    TG_POOL_ALLOC(s, &d_temp, (int64_t)temp_bytes);
    hipError_t e = rocprim::radix_sort_pairs(d_temp, temp_bytes, d_keys, d_keys2,
                                             d_vals, d_vals2, (size_t)n, 0, bits,
                                             s->stream);
    if (e != hipSuccess) { TG_SET_ERR("rocprim radix_sort_pairs: %s", hipGetErrorName(e)); return TG_ERR_HIP; }
    TG_HIP_CHECK(hipMemcpyAsync(d_keys, d_keys2, n * 8, hipMemcpyDeviceToDevice, s->stream));
    TG_HIP_CHECK(hipMemcpyAsync(d_vals, d_vals2, n * 8, hipMemcpyDeviceToDevice, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_keys2);
    tg_pool_free(s, d_vals2);
    tg_pool_free(s, d_temp);
    return TG_OK;
}

tg_status run_sort_pairs(tg_session* s, uint64_t* d_keys, int64_t* d_vals, int64_t n)
{
    return run_sort_pairs_bits(s, d_keys, d_vals, n, 64);
}

/* stable ascending sort of (u32 key, i32 value) pairs (join match
 * reordering: sort matches by probe row) */
tg_status run_sort_pairs_u32(tg_session* s, uint32_t* d_keys, int32_t* d_vals, int64_t n)
{
    if (n <= 1) return TG_OK;
    uint32_t* d_keys2 = nullptr;
    int32_t* d_vals2 = nullptr;
    TG_POOL_ALLOC(s, &d_keys2, n * 4);
    TG_POOL_ALLOC(s, &d_vals2, n * 4);
    size_t temp_bytes = 0;
    hipError_t e0 = rocprim::radix_sort_pairs(nullptr, temp_bytes, d_keys, d_keys2,
                                              d_vals, d_vals2, (size_t)n, 0, 32,
                                              s->stream);
    if (e0 != hipSuccess) { TG_SET_ERR("rocprim size query: %s", hipGetErrorName(e0)); return TG_ERR_HIP; }
    void* d_temp = nullptr;
    TG_POOL_ALLOC(s, &d_temp, (int64_t)temp_bytes);
    hipError_t e = rocprim::radix_sort_pairs(d_temp, temp_bytes, d_keys, d_keys2,
                                             d_vals, d_vals2, (size_t)n, 0, 32,
                                             s->stream);
    if (e != hipSuccess) { TG_SET_ERR("rocprim radix_sort_pairs: %s", hipGetErrorName(e)); return TG_ERR_HIP; }
    TG_HIP_CHECK(hipMemcpyAsync(d_keys, d_keys2, n * 4, hipMemcpyDeviceToDevice, s->stream));
    TG_HIP_CHECK(hipMemcpyAsync(d_vals, d_vals2, n * 4, hipMemcpyDeviceToDevice, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_keys2);
    tg_pool_free(s, d_vals2);
    tg_pool_free(s, d_temp);
    return TG_OK;
}

/* argsort of non-negative i64 keys (ascending, stable): writes the
 * permutation as int32 into d_out_idx[n]. d_keys is clobbered. */
tg_status run_argsort_i64(tg_session* s, int64_t* d_keys, int64_t n, int32_t* d_out_idx)
{
    int64_t* d_vals = nullptr;
    TG_POOL_ALLOC(s, &d_vals, (n ? n : 1) * 8);
    hipLaunchKernelGGL(k_iota_i64, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                       d_vals, n);
    TG_HIP_CHECK(hipGetLastError());
    tg_status st = run_sort_pairs(s, (uint64_t*)d_keys, d_vals, n);
    if (st != TG_OK) return st;
    hipLaunchKernelGGL(k_i64_to_i32, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                       d_vals, n, d_out_idx);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_vals);
    return TG_OK;
}

/* sort-based DISTINCT of a non-negative BIGINT column: radix_sort_keys
 * + rocprim::unique compaction, sequential-bandwidth only. Replaces the
 * generic hash aggregation dedup when nearly every row is unique (Q16's
 * ~12M distinct (combo,suppkey) pairs: the keystore insert + group-remap
 * sort pay random HBM round-trips per row; the sorted path streams).
 * d_out must hold n entries; *h_out_n gets the distinct count. `bits` is
 * the key width to sort (<=64; fewer radix passes for packed keys). */
extern "C" tg_status tg_dedup_i64(tg_session* s, const int64_t* d_in,
    int64_t n, int32_t bits, int64_t* d_out, int64_t* h_out_n)
{
    if (!s || !d_in || !d_out || !h_out_n || n < 0 || bits < 1 || bits > 64) {
        TG_SET_ERR("invalid dedup spec");
        return TG_ERR_INVALID_ARG;
    }
    if (n <= 1) {
        if (n == 1)
            TG_HIP_CHECK(hipMemcpyAsync(d_out, d_in, 8,
                                        hipMemcpyDeviceToDevice, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        *h_out_n = n;
        return TG_OK;
    }
    uint64_t* d_a = nullptr;
    uint64_t* d_b = nullptr;
    TG_POOL_ALLOC(s, &d_a, n * 8);
    TG_POOL_ALLOC(s, &d_b, n * 8);
    TG_HIP_CHECK(hipMemcpyAsync(d_a, d_in, n * 8, hipMemcpyDeviceToDevice,
                                s->stream));
    size_t temp_bytes = 0;
    hipError_t e0 = rocprim::radix_sort_keys(nullptr, temp_bytes, d_a, d_b,
                                             (size_t)n, 0, bits, s->stream);
    if (e0 != hipSuccess) { TG_SET_ERR("rocprim size query: %s", hipGetErrorName(e0)); return TG_ERR_HIP; }
    void* d_temp = nullptr;
    TG_POOL_ALLOC(s, &d_temp, (int64_t)temp_bytes);
    hipError_t e = rocprim::radix_sort_keys(d_temp, temp_bytes, d_a, d_b,
                                            (size_t)n, 0, bits, s->stream);
    if (e != hipSuccess) { TG_SET_ERR("rocprim radix_sort_keys: %s", hipGetErrorName(e)); return TG_ERR_HIP; }
    tg_pool_free(s, d_temp);
    size_t u_bytes = 0;
    int64_t* d_count = nullptr;
    TG_POOL_ALLOC(s, &d_count, 8);
    e0 = rocprim::unique(nullptr, u_bytes, d_b, (uint64_t*)d_out,
                         (int64_t*)d_count, (size_t)n,
                         rocprim::equal_to<uint64_t>(), s->stream);
    if (e0 != hipSuccess) { TG_SET_ERR("rocprim unique size query: %s", hipGetErrorName(e0)); return TG_ERR_HIP; }
    TG_POOL_ALLOC(s, &d_temp, (int64_t)u_bytes);
    e = rocprim::unique(d_temp, u_bytes, d_b, (uint64_t*)d_out,
                        (int64_t*)d_count, (size_t)n,
                        rocprim::equal_to<uint64_t>(), s->stream);
    if (e != hipSuccess) { TG_SET_ERR("rocprim unique: %s", hipGetErrorName(e)); return TG_ERR_HIP; }
    TG_HIP_CHECK(hipMemcpyAsync(h_out_n, d_count, 8, hipMemcpyDeviceToHost,
                                s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_a);
    tg_pool_free(s, d_b);
    tg_pool_free(s, d_temp);
    tg_pool_free(s, d_count);
    return TG_OK;
}

/* keys-only ascending u32 sort (segment-occurrence index for the indexed
 * pool LIKE; positions emitted by atomic append arrive unordered) */
tg_status run_sort_keys_u32(tg_session* s, uint32_t* d_keys, int64_t n)
{
    if (n <= 1) return TG_OK;
    uint32_t* d_keys2 = nullptr;
    TG_POOL_ALLOC(s, &d_keys2, n * 4);
    size_t temp_bytes = 0;
    hipError_t e0 = rocprim::radix_sort_keys(nullptr, temp_bytes, d_keys,
                                             d_keys2, (size_t)n, 0, 32,
                                             s->stream);
    if (e0 != hipSuccess) { TG_SET_ERR("rocprim size query: %s", hipGetErrorName(e0)); return TG_ERR_HIP; }
    void* d_temp = nullptr;
    TG_POOL_ALLOC(s, &d_temp, (int64_t)temp_bytes);
    hipError_t e = rocprim::radix_sort_keys(d_temp, temp_bytes, d_keys,
                                            d_keys2, (size_t)n, 0, 32,
                                            s->stream);
    if (e != hipSuccess) { TG_SET_ERR("rocprim radix_sort_keys: %s", hipGetErrorName(e)); return TG_ERR_HIP; }
    TG_HIP_CHECK(hipMemcpyAsync(d_keys, d_keys2, n * 4,
                                hipMemcpyDeviceToDevice, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_keys2);
    tg_pool_free(s, d_temp);
    return TG_OK;
}
