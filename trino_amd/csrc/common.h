/* common.h — internal helpers for libtrino_gpu (MI355X/gfx950 product path).
 * Host C++ above the C-ABI in include/trino_gpu.h. No CPU compute fallback:
 * every entry point fails loudly without a HIP device.
 */
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>

#include "../../include/trino_gpu.h"

/* thread-local last error string surfaced via tg_last_error() */
extern thread_local std::string tg_error_buf;

#define TG_SET_ERR(...) do { \
    char _b[512]; snprintf(_b, sizeof(_b), __VA_ARGS__); tg_error_buf = _b; } while (0)

#define TG_HIP_CHECK(call) do { \
    hipError_t _e = (call); \
    if (_e != hipSuccess) { \
        TG_SET_ERR("HIP error %s at %s:%d: %s", hipGetErrorName(_e), __FILE__, __LINE__, #call); \
        return TG_ERR_HIP; } } while (0)

#include <map>

struct tg_session {
    int device;
    hipStream_t stream;
    hipEvent_t ev_start, ev_stop;
    /* caching device-memory pool: per-step hipMalloc/hipFree of multi-GB
     * buffers costs O(10ms) each (page mapping); operators allocate through
     * the pool instead (measured: Q3 SF100 step 1208ms -> kernel-bound).
     * Single-threaded per session (one driver thread per operator chain). */
    std::multimap<size_t, void*> pool_free;   /* size -> buffer */
    std::map<void*, size_t> pool_sizes;       /* live + cached buffer sizes */
    size_t pool_bytes = 0;                    /* live + cached total */
    size_t mem_cap = 0;                       /* soft cap (92% of VRAM) */
    void* pin_buf = nullptr;                  /* pinned DtoH staging buffer */
    size_t pin_cap = 0;
};

tg_status tg_pool_alloc(tg_session* s, void** out, size_t bytes);
void tg_pool_free(tg_session* s, void* p);

#define TG_POOL_ALLOC(s, pp, bytes) do {     tg_status _st = tg_pool_alloc((s), (void**)(pp), (size_t)(bytes));     if (_st != TG_OK) return _st; } while (0)

/* launch geometry for memory-bound grid-stride kernels
 * (cdna_hip_programming.md Guideline 11: cap ~2048 blocks, grid-stride) */
constexpr int TG_BLOCK = 256;
constexpr int TG_MAX_BLOCKS = 2048;

static inline int tg_grid_for(int64_t items, int per_thread = 1)
{
    int64_t threads = (items + per_thread - 1) / per_thread;
    int64_t blocks = (threads + TG_BLOCK - 1) / TG_BLOCK;
    if (blocks > TG_MAX_BLOCKS) blocks = TG_MAX_BLOCKS;
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

#ifdef __HIPCC__
/* 128-bit unsigned accumulator as two u64 (device) */
struct u128 {
    unsigned long long lo, hi;
    __device__ __host__ u128() : lo(0), hi(0) {}
    __device__ inline void add(unsigned long long y)
    {
        lo += y;
        hi += (lo < y);
    }
    __device__ inline void add128(const u128& o)
    {
        lo += o.lo;
        hi += o.hi + (lo < o.lo);
    }
};
#endif
