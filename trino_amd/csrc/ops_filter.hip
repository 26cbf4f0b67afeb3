/* ops_filter.hip — columnar filter + projection over selection vectors.
 *
 * Mirrors:
 *  - ColumnarFilter.filterPositionsRange/List (sql/gen/columnar/
 *    ColumnarFilter.java:42-52): predicate -> int32 selection vector,
 *    branch-free compaction (CallColumnarFilterGenerator.java:160-198:
 *    outputPositions[count]=position; count += result) — here wave ballot +
 *    prefix-sum + stable two-phase scatter (order-preserving);
 *  - NULL semantics: any NULL argument rejects the row (nullable
 *    specialization, CallColumnarFilterGenerator.java:160-198);
 *  - PageProjection.project over SelectedPositions (project/PageProcessor.
 *    java:274-305); per-query bytecode gen is replaced by a device postfix
 *    interpreter over the tg_expr IR (stack depth <= 4, checked at compile).
 */
#include "operators.h"
#include <cstdlib>

struct KCol { const void* data; const uint64_t* valid; int32_t type; int32_t _pad; };

/* fused dynamic filter (sql/gen/columnar/DynamicPageFilter.java analog):
 * membership bitmap over a dense build-key range, tested INSIDE the scan
 * kernels after the static predicate passes (no extra materialized pass) */
struct DF { const uint64_t* bm; int64_t mn, mx; int32_t col; int32_t _pad; };

__device__ static inline bool df_test(const DF& df, const KCol* cols, int64_t i)
{
    if (!df.bm) return true;
    const KCol& c = cols[df.col];
    if (c.valid && !((c.valid[i >> 6] >> (i & 63)) & 1)) return false;
    int64_t k = (c.type == TG_BIGINT) ? ((const int64_t*)c.data)[i]
                                      : (int64_t)((const int32_t*)c.data)[i];
    if (k < df.mn || k > df.mx) return false;
    int64_t b = k - df.mn;
    return (df.bm[b >> 6] >> (b & 63)) & 1;
}

#define MAX_STACK 6

/* Typed stack value: BIGINT/INTEGER/DATE/… stay in an exact int64 lane
 * (mirrors sql/gen/columnar/CallColumnarFilterGenerator.java:160-198 which
 * compiles the comparison at the column's Java type — long compares are
 * exact there, so they must be exact here; keys above 2^53 would silently
 * lose precision in a double lane). `f` mirrors the value for mixed-type
 * ops (Trino coerces BIGINT to DOUBLE in mixed expressions). */
struct TVal { double f; int64_t i; bool isint; bool null; };

__device__ static inline TVal load_tcol(const KCol& c, int64_t i)
{
    TVal v;
    v.null = c.valid && !((c.valid[i >> 6] >> (i & 63)) & 1);
    if (v.null) { v.f = 0.0; v.i = 0; v.isint = true; return v; }
    switch (c.type) {
        case TG_BIGINT: v.i = ((const int64_t*)c.data)[i]; v.isint = true; v.f = (double)v.i; break;
        case TG_INTEGER: case TG_DATE: v.i = ((const int32_t*)c.data)[i]; v.isint = true; v.f = (double)v.i; break;
        case TG_SMALLINT: v.i = ((const int16_t*)c.data)[i]; v.isint = true; v.f = (double)v.i; break;
        case TG_TINYINT: case TG_BOOLEAN: v.i = ((const int8_t*)c.data)[i]; v.isint = true; v.f = (double)v.i; break;
        default: v.f = ((const double*)c.data)[i]; v.i = 0; v.isint = false; break;
    }
    return v;
}

__device__ static inline bool tval_truthy(const TVal& v) { return v.isint ? v.i != 0 : v.f != 0.0; }

/* postfix interpreter; returns a typed value. Integer ops are exact int64
 * (incl. BIGINT/BIGINT truncated division); mixed int/double coerces to
 * double. Comparisons compare in int64 iff both sides are int. AND/OR/NOT
 * follow SQL Kleene three-valued logic (io.trino.sql.ir.Logical semantics:
 * AND — false dominates null; OR — true dominates null; NOT null = null),
 * so compositions under NOT are correct; a null FILTER result rejects the
 * row at the call site (CallColumnarFilterGenerator's nullable loop). */
__device__ static TVal eval_expr(const tg_expr_inst* prog, int count,
                                 const KCol* cols, int64_t row)
{
    TVal s0{}, s1{}, s2{}, s3{}, s4{}, s5{};
    int sp = 0;
#define PUSH(v) do { switch (sp) { \
        case 0: s0 = (v); break; case 1: s1 = (v); break; \
        case 2: s2 = (v); break; case 3: s3 = (v); break; \
        case 4: s4 = (v); break; default: s5 = (v); break; } \
        sp++; } while (0)
#define TOP (sp == 1 ? s0 : sp == 2 ? s1 : sp == 3 ? s2 : sp == 4 ? s3 \
             : sp == 5 ? s4 : s5)
#define POP(v) do { v = TOP; sp--; } while (0)
#define ARITH(expr_i, expr_f) do { TVal r; r.null = a.null | b.null; \
        r.isint = a.isint & b.isint; \
        if (r.isint) { r.i = (expr_i); r.f = (double)r.i; } \
        else { r.f = (expr_f); r.i = 0; } PUSH(r); } while (0)
#define CMP(op_) do { TVal r; r.null = a.null | b.null; r.isint = true; \
        r.i = (a.isint & b.isint) ? (a.i op_ b.i ? 1 : 0) \
                                  : (a.f op_ b.f ? 1 : 0); \
        r.f = (double)r.i; PUSH(r); } while (0)

    for (int k = 0; k < count; k++) {
        tg_expr_inst in = prog[k];
        TVal a, b, c;
        switch (in.op) {
            case TG_EXPR_COL: PUSH(load_tcol(cols[in.arg0], row)); break;
            case TG_EXPR_CONST_F64: { TVal v; v.f = in.imm.f64; v.i = 0; v.isint = false; v.null = false; PUSH(v); break; }
            case TG_EXPR_CONST_I64: { TVal v; v.i = in.imm.i64; v.f = (double)v.i; v.isint = true; v.null = false; PUSH(v); break; }
            case TG_EXPR_ADD: POP(b); POP(a); ARITH(a.i + b.i, a.f + b.f); break;
            case TG_EXPR_SUB: POP(b); POP(a); ARITH(a.i - b.i, a.f - b.f); break;
            case TG_EXPR_MUL: POP(b); POP(a); ARITH(a.i * b.i, a.f * b.f); break;
            case TG_EXPR_DIV: POP(b); POP(a); {
                TVal r; r.isint = a.isint & b.isint;
                if (r.isint) {   /* BIGINT/BIGINT truncates; /0 poisons to null
                                    (the C-ABI has no per-row error channel;
                                    Trino raises DIVISION_BY_ZERO) */
                    r.null = a.null | b.null | (b.i == 0);
                    r.i = (b.i == 0) ? 0 : a.i / b.i;
                    r.f = (double)r.i;
                }
                else { r.null = a.null | b.null; r.f = a.f / b.f; r.i = 0; }
                PUSH(r); break;
            }
            case TG_EXPR_LE: POP(b); POP(a); CMP(<=); break;
            case TG_EXPR_LT: POP(b); POP(a); CMP(<); break;
            case TG_EXPR_GE: POP(b); POP(a); CMP(>=); break;
            case TG_EXPR_GT: POP(b); POP(a); CMP(>); break;
            case TG_EXPR_EQ: POP(b); POP(a); CMP(==); break;
            case TG_EXPR_NE: POP(b); POP(a); CMP(!=); break;
            case TG_EXPR_AND: { POP(b); POP(a);
                bool af = !a.null && !tval_truthy(a), bf = !b.null && !tval_truthy(b);
                TVal r; r.isint = true;
                r.null = !(af | bf) && (a.null | b.null);
                r.i = (!r.null && !af && !bf) ? 1 : 0;
                r.f = (double)r.i; PUSH(r); break;
            }
            case TG_EXPR_OR: { POP(b); POP(a);
                bool at = !a.null && tval_truthy(a), bt = !b.null && tval_truthy(b);
                TVal r; r.isint = true;
                r.null = !(at | bt) && (a.null | b.null);
                r.i = (at | bt) ? 1 : 0;
                r.f = (double)r.i; PUSH(r); break;
            }
            case TG_EXPR_NOT: { POP(a);
                TVal r; r.isint = true; r.null = a.null;
                r.i = (!a.null && !tval_truthy(a)) ? 1 : 0;
                r.f = (double)r.i; PUSH(r); break;
            }
            case TG_EXPR_BETWEEN: { POP(c); POP(b); POP(a);
                /* a >= b AND a <= c under Kleene AND, typed compares */
                bool ge, le, ge_n = a.null | b.null, le_n = a.null | c.null;
                ge = (a.isint & b.isint) ? a.i >= b.i : a.f >= b.f;
                le = (a.isint & c.isint) ? a.i <= c.i : a.f <= c.f;
                bool af = !ge_n && !ge, bf = !le_n && !le;
                TVal r; r.isint = true;
                r.null = !(af | bf) && (ge_n | le_n);
                r.i = (!r.null && !af && !bf) ? 1 : 0;
                r.f = (double)r.i; PUSH(r); break;
            }
            default: break;
        }
    }
    return TOP;
#undef PUSH
#undef POP
#undef TOP
#undef ARITH
#undef CMP
}

/* row index for the k-th selected input position */
__device__ static inline int64_t sel_row(int has_list, const int32_t* list,
                                         int32_t offset, int64_t k)
{
    return has_list ? (int64_t)list[k] : offset + k;
}

/* specialization for the dominant shape COL <cmp> CONST (ColumnarFilter's
 * compiled single-predicate case): no interpreter loop, vector-friendly.
 * int_mode: col is an integer type and the constant was CONST_I64 —
 * compare exactly in int64 (keys above 2^53 must not round). */
__global__ void k_filter_cmp(KCol col, int op, double cval, int64_t icval, int int_mode,
                             int has_list, const int32_t* list, int32_t offset,
                             int64_t n, uint8_t* __restrict__ flags,
                             DF df, const KCol* __restrict__ all_cols)
{
    int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; k < n; k += stride) {
        int64_t i = has_list ? (int64_t)list[k] : offset + k;
        bool r;
        TVal v = load_tcol(col, i);
        if (int_mode) {
            switch (op) {
                case TG_EXPR_LE: r = v.i <= icval; break;
                case TG_EXPR_LT: r = v.i < icval; break;
                case TG_EXPR_GE: r = v.i >= icval; break;
                case TG_EXPR_GT: r = v.i > icval; break;
                case TG_EXPR_EQ: r = v.i == icval; break;
                default: r = v.i != icval; break;
            }
        }
        else {
            switch (op) {
                case TG_EXPR_LE: r = v.f <= cval; break;
                case TG_EXPR_LT: r = v.f < cval; break;
                case TG_EXPR_GE: r = v.f >= cval; break;
                case TG_EXPR_GT: r = v.f > cval; break;
                case TG_EXPR_EQ: r = v.f == cval; break;
                default: r = v.f != cval; break;
            }
        }
        bool pass = !v.null && r;
        if (pass) pass = df_test(df, all_cols, i);
        flags[k] = pass ? 1 : 0;
    }
}

/* conjunction fast path: AND-chains of simple terms (the dominant sweep
 * filter shape — Q4/Q6/Q12's date windows, col-col compares and IN pairs).
 * Extracted host-side from the postfix program; each term is evaluated
 * directly, early-out on the first false, no interpreter stack. */
#define FTERM_MAX 8
struct FTerm {
    int32_t kind;     /* 0 = colA cmp const, 1 = colA cmp colB,
                         2 = colA between [c1,c2], 3 = colA in {c1,c2} */
    int32_t colA, colB, op;
    double c1, c2;
    int64_t i1, i2;   /* exact int64 views of c1/c2 when int_mode */
    int32_t int_mode; /* colA is an integer type and consts were CONST_I64 */
    int32_t _pad;
};
struct FTerms { FTerm t[FTERM_MAX]; int n; };

__device__ static inline bool fterm_cmp(int op, double a, double b)
{
    switch (op) {
        case TG_EXPR_LE: return a <= b;
        case TG_EXPR_LT: return a < b;
        case TG_EXPR_GE: return a >= b;
        case TG_EXPR_GT: return a > b;
        case TG_EXPR_EQ: return a == b;
        default: return a != b;
    }
}

__device__ static inline bool fterm_cmpi(int op, int64_t a, int64_t b)
{
    switch (op) {
        case TG_EXPR_LE: return a <= b;
        case TG_EXPR_LT: return a < b;
        case TG_EXPR_GE: return a >= b;
        case TG_EXPR_GT: return a > b;
        case TG_EXPR_EQ: return a == b;
        default: return a != b;
    }
}

__global__ void k_filter_terms(FTerms ft, const KCol* __restrict__ cols,
                               int has_list, const int32_t* __restrict__ list,
                               int32_t offset, int64_t n,
                               uint8_t* __restrict__ flags, DF df)
{
    int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; k < n; k += stride) {
        int64_t i = has_list ? (int64_t)list[k] : offset + k;
        bool pass = true;
        for (int t = 0; t < ft.n && pass; t++) {
            const FTerm& f = ft.t[t];
            TVal a = load_tcol(cols[f.colA], i);
            switch (f.kind) {
                case 0: pass = !a.null && (f.int_mode ? fterm_cmpi(f.op, a.i, f.i1)
                                                      : fterm_cmp(f.op, a.f, f.c1)); break;
                case 1: {
                    TVal b = load_tcol(cols[f.colB], i);
                    pass = !a.null && !b.null &&
                           ((a.isint & b.isint) ? fterm_cmpi(f.op, a.i, b.i)
                                                : fterm_cmp(f.op, a.f, b.f));
                    break;
                }
                case 2: pass = !a.null && (f.int_mode ? (a.i >= f.i1 && a.i <= f.i2)
                                                      : (a.f >= f.c1 && a.f <= f.c2)); break;
                default: pass = !a.null && (f.int_mode ? (a.i == f.i1 || a.i == f.i2)
                                                       : (a.f == f.c1 || a.f == f.c2)); break;
            }
        }
        if (pass) pass = df_test(df, cols, i);
        flags[k] = pass ? 1 : 0;
    }
}

/* try to parse the postfix program as T1 (T2 AND)*; returns term count or 0 */
static bool type_is_int(int32_t t)
{
    return t == TG_BIGINT || t == TG_INTEGER || t == TG_DATE ||
           t == TG_SMALLINT || t == TG_TINYINT || t == TG_BOOLEAN;
}

static int parse_fterms(const ExprProgram& pred, const std::vector<KCol>& cols, FTerms* out)
{
    /* i64-ness tracked per constant so integer-column terms compare exactly */
    auto cval = [](const tg_expr_inst& in, double* v, int64_t* iv, bool* is_i64) {
        if (in.op == TG_EXPR_CONST_F64) { *v = in.imm.f64; *iv = (int64_t)in.imm.f64; *is_i64 = false; return true; }
        if (in.op == TG_EXPR_CONST_I64) { *v = (double)in.imm.i64; *iv = in.imm.i64; *is_i64 = true; return true; }
        return false;
    };
    auto is_cmp = [](int op) { return op >= TG_EXPR_LE && op <= TG_EXPR_NE; };
    int p = 0, nt = 0;
    const auto* I = pred.insts.data();
    int N = pred.count;
    auto term = [&]() -> bool {
        if (nt >= FTERM_MAX || p >= N || I[p].op != TG_EXPR_COL) return false;
        FTerm& f = out->t[nt];
        bool i1 = false, i2 = false;
        bool col_int = type_is_int(cols[I[p].arg0].type);
        /* IN-pair: COL C EQ COL C EQ OR (same column) */
        if (p + 6 < N && cval(I[p + 1], &f.c1, &f.i1, &i1) && I[p + 2].op == TG_EXPR_EQ &&
            I[p + 3].op == TG_EXPR_COL && I[p + 3].arg0 == I[p].arg0 &&
            cval(I[p + 4], &f.c2, &f.i2, &i2) && I[p + 5].op == TG_EXPR_EQ &&
            I[p + 6].op == TG_EXPR_OR) {
            f.kind = 3; f.colA = I[p].arg0; f.colB = -1; f.op = 0;
            f.int_mode = col_int && i1 && i2;
            p += 7; nt++; return true;
        }
        /* BETWEEN: COL C C BETWEEN */
        if (p + 3 < N && cval(I[p + 1], &f.c1, &f.i1, &i1) &&
            cval(I[p + 2], &f.c2, &f.i2, &i2) &&
            I[p + 3].op == TG_EXPR_BETWEEN) {
            f.kind = 2; f.colA = I[p].arg0; f.colB = -1; f.op = 0;
            f.int_mode = col_int && i1 && i2;
            p += 4; nt++; return true;
        }
        /* COL cmp CONST */
        if (p + 2 < N && cval(I[p + 1], &f.c1, &f.i1, &i1) && is_cmp(I[p + 2].op)) {
            f.kind = 0; f.colA = I[p].arg0; f.colB = -1; f.op = I[p + 2].op;
            f.int_mode = col_int && i1;
            p += 3; nt++; return true;
        }
        /* COL cmp COL */
        if (p + 2 < N && I[p + 1].op == TG_EXPR_COL && is_cmp(I[p + 2].op)) {
            f.kind = 1; f.colA = I[p].arg0; f.colB = I[p + 1].arg0;
            f.op = I[p + 2].op; f.c1 = f.c2 = 0; f.i1 = f.i2 = 0; f.int_mode = 0;
            p += 3; nt++; return true;
        }
        return false;
    };
    if (!term()) return 0;
    while (p < N) {
        if (!term()) return 0;
        if (p >= N || I[p].op != TG_EXPR_AND) return 0;
        p++;
    }
    out->n = nt;
    return nt;
}

__global__ void k_filter_flags(const tg_expr_inst* prog, int count, const KCol* cols,
                               int has_list, const int32_t* list, int32_t offset,
                               int64_t n, uint8_t* __restrict__ flags, DF df)
{
    int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; k < n; k += stride) {
        int64_t i = sel_row(has_list, list, offset, k);
        bool pass;
        if (count > 0) {
            TVal v = eval_expr(prog, count, cols, i);
            pass = !v.null && tval_truthy(v);
        }
        else {
            pass = true;    /* dynamic-filter-only scan */
        }
        if (pass) pass = df_test(df, cols, i);
        flags[k] = pass ? 1 : 0;
    }
}

/* stable compaction: per-chunk counts -> scan -> scatter. Chunks are
 * CHUNK-sized contiguous ranges so ordering is preserved. */
#define CHUNK 16384

__global__ void k_count_chunk(const uint8_t* __restrict__ flags, int64_t n,
                              int32_t* __restrict__ chunk_counts, int64_t nchunks)
{
    /* one wave per chunk, lanes stride u64 words (8 flag bytes each, flags
     * are 0x00/0x01 so popcount(word) == byte sum): coalesced, ~full HBM
     * rate. The earlier thread-per-chunk serial byte loop ran at 0.26 TB/s. */
    int64_t c = (int64_t)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    if (c >= nchunks) return;
    int lane = threadIdx.x % 64;
    int64_t lo = c * CHUNK, hi = min(lo + CHUNK, n);
    int32_t cnt = 0;
    int64_t w_lo = lo / 8, w_hi = hi / 8;   /* CHUNK and n8 boundaries */
    const uint64_t* w = (const uint64_t*)flags;
    for (int64_t k = w_lo + lane; k < w_hi; k += 64) cnt += __popcll(w[k]);
    for (int64_t i = w_hi * 8 + lane; i < hi; i += 64) cnt += flags[i];
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1) cnt += __shfl_xor(cnt, off, 64);
    if (lane == 0) chunk_counts[c] = cnt;
}

/* hierarchical in-place i32 exclusive scan: wave-per-chunk sums, serial
 * chunk scan (chunk count is small), wave-per-chunk offsets. The one-wave
 * k_scan_serial costs ~1.5 ms at 37k entries (3x per Q3 step); this runs in
 * ~tens of µs and scales to multi-million-entry scans (varchar gathers). */
#define FSCAN_CHUNK 2048

__global__ void k_s32_chunk_sums(const int32_t* __restrict__ arr, int64_t n,
                                 int32_t* __restrict__ chunk_sums, int64_t nchunks)
{
    int64_t c = (int64_t)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    if (c >= nchunks) return;
    int lane = threadIdx.x % 64;
    int64_t lo = c * FSCAN_CHUNK, hi = min(lo + FSCAN_CHUNK, n);
    int32_t s = 0;
    for (int64_t i = lo + lane; i < hi; i += 64) s += arr[i];
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1) s += __shfl_xor(s, off, 64);
    if (lane == 0) chunk_sums[c] = s;
}

__global__ void k_s32_chunks_serial(int32_t* chunk_sums, int64_t nchunks, int32_t* total)
{
    if (blockIdx.x || threadIdx.x) return;
    int32_t run = 0;
    for (int64_t i = 0; i < nchunks; i++) {
        int32_t v = chunk_sums[i];
        chunk_sums[i] = run;
        run += v;
    }
    if (total) *total = run;
}

__global__ void k_s32_offsets(int32_t* __restrict__ arr, int64_t n,
                              const int32_t* __restrict__ chunk_sums, int64_t nchunks)
{
    int64_t c = (int64_t)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    if (c >= nchunks) return;
    int lane = threadIdx.x % 64;
    int64_t lo = c * FSCAN_CHUNK, hi = min(lo + FSCAN_CHUNK, n);
    int32_t run = chunk_sums[c];
    for (int64_t g = lo; g < hi; g += 64) {
        int64_t i = g + lane;
        int32_t v = (i < hi) ? arr[i] : 0;
        int32_t pre = v;
        #pragma unroll
        for (int off = 1; off < 64; off <<= 1) {
            int32_t o = __shfl_up(pre, off, 64);
            if (lane >= off) pre += o;
        }
        int32_t wave_total = __shfl(pre, 63, 64);
        if (i < hi) arr[i] = run + pre - v;
        run += wave_total;
    }
}

__global__ void k_scan_serial(int32_t* counts, int64_t n, int32_t* total)
{
    /* one-wave exclusive scan: 64-wide segments with a carried base
     * (the former one-thread loop cost ~1.5 ms at 37k chunks) */
    if (blockIdx.x || threadIdx.x >= 64) return;
    int lane = threadIdx.x;
    int32_t run = 0;
    for (int64_t g = 0; g < n; g += 64) {
        int64_t i = g + lane;
        int32_t v = (i < n) ? counts[i] : 0;
        int32_t pre = v;
        #pragma unroll
        for (int off = 1; off < 64; off <<= 1) {
            int32_t o = __shfl_up(pre, off, 64);
            if (lane >= off) pre += o;
        }
        int32_t seg_total = __shfl(pre, 63, 64);
        if (i < n) counts[i] = run + pre - v;
        run += seg_total;
    }
    if (lane == 0) *total = run;
}

tg_status run_scan_i32(tg_session* s, int32_t* d_arr, int64_t n, int32_t* d_total)
{
    if (n <= 2 * FSCAN_CHUNK) {   /* small: one launch beats three */
        hipLaunchKernelGGL(k_scan_serial, dim3(1), dim3(64), 0, s->stream,
                           d_arr, n, d_total);
        TG_HIP_CHECK(hipGetLastError());
        return TG_OK;
    }
    int64_t nchunks = (n + FSCAN_CHUNK - 1) / FSCAN_CHUNK;
    int32_t* d_cs = nullptr;
    TG_POOL_ALLOC(s, &d_cs, nchunks * 4);
    int wpb = TG_BLOCK / 64;
    hipLaunchKernelGGL(k_s32_chunk_sums, dim3((uint32_t)((nchunks + wpb - 1) / wpb)),
                       dim3(TG_BLOCK), 0, s->stream, d_arr, n, d_cs, nchunks);
    TG_HIP_CHECK(hipGetLastError());
    /* the middle scan recurses: at 600M inputs nchunks is ~293k and a
     * single-thread pass costs ~15 ms */
    tg_status st = run_scan_i32(s, d_cs, nchunks, d_total);
    if (st != TG_OK) return st;
    hipLaunchKernelGGL(k_s32_offsets, dim3((uint32_t)((nchunks + wpb - 1) / wpb)),
                       dim3(TG_BLOCK), 0, s->stream, d_arr, n, d_cs, nchunks);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_cs);
    return TG_OK;
}

__global__ void k_compact(const uint8_t* __restrict__ flags, int64_t n,
                          const int32_t* __restrict__ chunk_offsets,
                          int has_list, const int32_t* list, int32_t offset,
                          int32_t* __restrict__ out_positions)
{
    /* one WAVE per chunk walks it in 64-wide steps carrying a running
     * count — order within the chunk (and via the scanned chunk offsets,
     * globally) is preserved */
    int64_t c = blockIdx.x;
    int64_t lo = c * CHUNK, hi = min(lo + CHUNK, n);
    int lane = threadIdx.x % 64;
    if (threadIdx.x >= 64) return;
    int32_t run = chunk_offsets[c];
    for (int64_t g = lo; g < hi; g += 64) {
        int64_t i = g + lane;
        bool f = (i < hi) && flags[i];
        unsigned long long b = __ballot(f);
        int before = __popcll(b & ((1ull << lane) - 1ull));
        if (f) {
            int64_t k = i;
            out_positions[run + before] =
                has_list ? list[k] : (int32_t)(offset + k);
        }
        run += __popcll(b);
    }
}

/* ---- fused filter+project (the ScanFilterAndProjectOperator shape:
 * PageProcessor evaluates filter then projections over the selected
 * positions in one operator; here two passes with no materialized selection
 * vector: pass 1 counts survivors per chunk, pass 2 re-evaluates the
 * predicate and writes every projection at the scanned offsets) ---- */
struct KProj {
    const tg_expr_inst* insts;
    int32_t count;
    int32_t identity_col;   /* >=0: typed copy of that column */
    int32_t out_type;
    void* out;
    uint64_t* out_valid;
};
#define MAX_PROJ 8

__global__ void k_fp_count(const tg_expr_inst* prog, int count, const KCol* cols,
                           int64_t n, int32_t* __restrict__ chunk_counts, int64_t nchunks)
{
    /* one BLOCK per chunk; thread-per-row evaluation, block-reduced count */
    int64_t c = blockIdx.x;
    if (c >= nchunks) return;
    int64_t lo = c * CHUNK, hi = min(lo + CHUNK, n);
    int32_t cnt = 0;
    for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        TVal v = eval_expr(prog, count, cols, i);
        cnt += (!v.null && tval_truthy(v)) ? 1 : 0;
    }
    #pragma unroll
    for (int off = 32; off >= 1; off >>= 1) cnt += __shfl_xor(cnt, off, 64);
    __shared__ int32_t wsum[TG_BLOCK / 64];
    if (threadIdx.x % 64 == 0) wsum[threadIdx.x / 64] = cnt;
    __syncthreads();
    if (threadIdx.x == 0) {
        int32_t t = 0;
        for (int w = 0; w < TG_BLOCK / 64; w++) t += wsum[w];
        chunk_counts[c] = t;
    }
}

__global__ void k_fp_write(const tg_expr_inst* prog, int count, const KCol* cols,
                           int64_t n, const int32_t* __restrict__ chunk_offsets,
                           const KProj* projs, int nproj, int64_t nchunks)
{
    /* one BLOCK per chunk: flags staged as an LDS bitmask, per-wave
     * sub-range offsets, then 64-wide selected-only projection evaluation */
    int64_t c = blockIdx.x;
    if (c >= nchunks) return;
    constexpr int WAVES = TG_BLOCK / 64;
    constexpr int GROUPS = CHUNK / 64;          /* 64-row groups per chunk */
    __shared__ unsigned long long mask[GROUPS];
    __shared__ int32_t wave_base[WAVES];
    int64_t lo = c * CHUNK, hi = min(lo + CHUNK, n);
    int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
    /* phase 1: evaluate predicate, ballot into the LDS mask */
    for (int g = wave; g < GROUPS; g += WAVES) {
        int64_t i = lo + (int64_t)g * 64 + lane;
        bool sel = false;
        if (i < hi) {
            TVal v = eval_expr(prog, count, cols, i);
            sel = !v.null && tval_truthy(v);
        }
        unsigned long long b = __ballot(sel);
        if (lane == 0) mask[g] = b;
    }
    __syncthreads();
    /* phase 2: per-wave quarter counts -> exclusive wave bases */
    constexpr int GPW = GROUPS / WAVES;         /* groups per wave */
    {
        int32_t cnt = 0;
        for (int g = wave * GPW + lane; g < (wave + 1) * GPW; g += 64)
            cnt += __popcll(mask[g]);
        #pragma unroll
        for (int off = 32; off >= 1; off >>= 1) cnt += __shfl_xor(cnt, off, 64);
        if (lane == 0) wave_base[wave] = cnt;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        int32_t run = 0;
        for (int w = 0; w < WAVES; w++) {
            int32_t v = wave_base[w];
            wave_base[w] = run;
            run += v;
        }
    }
    __syncthreads();
    /* phase 3: selected lanes evaluate projections and write compacted */
    int32_t run = chunk_offsets[c] + wave_base[wave];
    for (int g = wave * GPW; g < (wave + 1) * GPW; g++) {
        unsigned long long b = mask[g];
        int64_t i = lo + (int64_t)g * 64 + lane;
        bool sel = (b >> lane) & 1;
        if (sel) {
            int before = __popcll(b & ((1ull << lane) - 1ull));
            int64_t at = run + before;
            for (int p = 0; p < nproj; p++) {
                const KProj& pr = projs[p];
                if (pr.identity_col >= 0) {
                    const KCol& sc = cols[pr.identity_col];
                    switch (pr.out_type) {
                        case TG_BIGINT: ((int64_t*)pr.out)[at] = ((const int64_t*)sc.data)[i]; break;
                        case TG_INTEGER: case TG_DATE: ((int32_t*)pr.out)[at] = ((const int32_t*)sc.data)[i]; break;
                        case TG_SMALLINT: ((int16_t*)pr.out)[at] = ((const int16_t*)sc.data)[i]; break;
                        case TG_TINYINT: case TG_BOOLEAN: ((int8_t*)pr.out)[at] = ((const int8_t*)sc.data)[i]; break;
                        default: ((double*)pr.out)[at] = ((const double*)sc.data)[i]; break;
                    }
                    if (pr.out_valid && sc.valid && !((sc.valid[i >> 6] >> (i & 63)) & 1))
                        atomicAnd((unsigned long long*)&pr.out_valid[at >> 6],
                                  ~(1ull << (at & 63)));
                }
                else {
                    TVal pv = eval_expr(pr.insts, pr.count, cols, i);
                    if (pr.out_type == TG_BIGINT)
                        ((int64_t*)pr.out)[at] = pv.null ? 0
                            : (pv.isint ? pv.i : (int64_t)pv.f);
                    else
                        ((double*)pr.out)[at] = pv.null ? 0.0 : pv.f;
                    if (pr.out_valid && pv.null)
                        atomicAnd((unsigned long long*)&pr.out_valid[at >> 6],
                                  ~(1ull << (at & 63)));
                }
            }
        }
        run += __popcll(b);
    }
}

tg_status run_filter(tg_session* s, const ExprProgram& pred, const DevPage& page,
                     const tg_selected* input_sel,
                     int32_t** d_positions_out, int32_t* count_out,
                     const DF* df_in)
{
    DF df{nullptr, 0, 0, 0, 0};
    if (df_in) df = *df_in;
    /* device column descriptors */
    std::vector<KCol> cols(page.blocks.size());
    for (size_t i = 0; i < page.blocks.size(); i++)
        cols[i] = {page.blocks[i].data, page.blocks[i].valid, (int32_t)page.blocks[i].type, 0};
    KCol* d_cols = nullptr;
    TG_POOL_ALLOC(s, &d_cols, cols.size() * sizeof(KCol));
    TG_HIP_CHECK(hipMemcpyAsync(d_cols, cols.data(), cols.size() * sizeof(KCol),
                                hipMemcpyHostToDevice, s->stream));

    int has_list = input_sel && input_sel->is_list;
    int32_t offset = input_sel ? input_sel->offset : 0;
    int64_t n = input_sel ? input_sel->size : page.n;
    const int32_t* d_list = nullptr;
    int32_t* d_list_owned = nullptr;
    if (has_list) {
        TG_POOL_ALLOC(s, &d_list_owned, n * sizeof(int32_t));
        TG_HIP_CHECK(hipMemcpyAsync(d_list_owned, input_sel->positions, n * sizeof(int32_t),
                                    hipMemcpyHostToDevice, s->stream));
        d_list = d_list_owned;
    }

    uint8_t* d_flags = nullptr;
    TG_POOL_ALLOC(s, &d_flags, n ? n : 1);
    /* fast path: [COL c][CONST v][cmp] */
    bool fast = pred.count == 3 && pred.insts[0].op == TG_EXPR_COL &&
                (pred.insts[1].op == TG_EXPR_CONST_F64 || pred.insts[1].op == TG_EXPR_CONST_I64) &&
                pred.insts[2].op >= TG_EXPR_LE && pred.insts[2].op <= TG_EXPR_NE;
    if (fast) {
        bool is_i64 = pred.insts[1].op == TG_EXPR_CONST_I64;
        double cval = is_i64 ? (double)pred.insts[1].imm.i64 : pred.insts[1].imm.f64;
        int64_t icval = is_i64 ? pred.insts[1].imm.i64 : (int64_t)pred.insts[1].imm.f64;
        int int_mode = is_i64 && type_is_int(cols[pred.insts[0].arg0].type);
        hipLaunchKernelGGL(k_filter_cmp, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                           cols[pred.insts[0].arg0], pred.insts[2].op, cval, icval, int_mode,
                           has_list, d_list, offset, n, d_flags, df, d_cols);
    }
    else if (FTerms ft{}; pred.count > 0 && parse_fterms(pred, cols, &ft) > 0) {
        hipLaunchKernelGGL(k_filter_terms, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                           ft, d_cols, has_list, d_list, offset, n, d_flags, df);
    }
    else {
        hipLaunchKernelGGL(k_filter_flags, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                           pred.d_insts, pred.count, d_cols, has_list, d_list, offset, n, d_flags, df);
    }
    TG_HIP_CHECK(hipGetLastError());

    int64_t nchunks = (n + CHUNK - 1) / CHUNK;
    if (nchunks < 1) nchunks = 1;
    int32_t* d_offsets = nullptr;
    int32_t* d_total = nullptr;
    TG_POOL_ALLOC(s, &d_offsets, nchunks * sizeof(int32_t));
    TG_POOL_ALLOC(s, &d_total, sizeof(int32_t));
    hipLaunchKernelGGL(k_count_chunk,
                       dim3((uint32_t)((nchunks + TG_BLOCK / 64 - 1) / (TG_BLOCK / 64))),
                       dim3(TG_BLOCK), 0, s->stream,
                       d_flags, n, d_offsets, nchunks);
    TG_HIP_CHECK(hipGetLastError());
    tg_status scst = run_scan_i32(s, d_offsets, nchunks, d_total);
    if (scst != TG_OK) return scst;
    int32_t total = 0;
    TG_HIP_CHECK(hipMemcpyAsync(&total, d_total, 4, hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));

    int32_t* d_pos = nullptr;
    TG_POOL_ALLOC(s, &d_pos, (total ? total : 1) * sizeof(int32_t));
    hipLaunchKernelGGL(k_compact, dim3((int)nchunks), dim3(64), 0, s->stream,
                       d_flags, n, d_offsets, has_list, d_list, offset, d_pos);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));

    tg_pool_free(s, d_flags);
    tg_pool_free(s, d_offsets);
    tg_pool_free(s, d_total);
    tg_pool_free(s, d_cols);
    if (d_list_owned) tg_pool_free(s, d_list_owned);
    *d_positions_out = d_pos;
    *count_out = total;
    return TG_OK;
}

/* ---- projection ---- */
__global__ void k_project(const tg_expr_inst* prog, int count, const KCol* cols,
                          const int32_t* __restrict__ positions, int32_t n,
                          void* __restrict__ out, int out_is_i64,
                          uint64_t* __restrict__ out_valid)
{
    int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; k < n; k += stride) {
        TVal v = eval_expr(prog, count, cols, positions ? positions[k] : k);
        if (out_is_i64)
            ((int64_t*)out)[k] = v.null ? 0 : (v.isint ? v.i : (int64_t)v.f);
        else
            ((double*)out)[k] = v.null ? 0.0 : v.f;
        if (out_valid && v.null)
            atomicAnd((unsigned long long*)&out_valid[k >> 6], ~(1ull << (k & 63)));
    }
}

template <typename T>
__global__ void k_gather(const T* __restrict__ src, const uint64_t* __restrict__ src_valid,
                         const int32_t* __restrict__ positions, int32_t n,
                         T* __restrict__ out, uint64_t* __restrict__ out_valid)
{
    int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; k < n; k += stride) {
        int64_t i = positions ? positions[k] : k;
        if (i < 0) {   /* probe-outer null position */
            out[k] = T();
            if (out_valid)
                atomicAnd((unsigned long long*)&out_valid[k >> 6], ~(1ull << (k & 63)));
            continue;
        }
        out[k] = src[i];
        if (out_valid && src_valid && !((src_valid[i >> 6] >> (i & 63)) & 1))
            atomicAnd((unsigned long long*)&out_valid[k >> 6], ~(1ull << (k & 63)));
    }
}

/* variable-width gather (VariableWidthBlock.copyPositions analog:
 * spi/block/VariableWidthBlock.java:171-198 — lengths pass, offset prefix,
 * byte copy pass) */
__global__ void k_gather_var_lens(const int32_t* __restrict__ src_off,
                                  const uint64_t* __restrict__ src_valid,
                                  const int32_t* __restrict__ pos, int32_t n,
                                  int32_t* __restrict__ lens,
                                  uint64_t* __restrict__ out_valid)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int32_t p = pos[i];
        bool isnull = p < 0 ||
                      (src_valid && !((src_valid[p >> 6] >> (p & 63)) & 1));
        lens[i] = isnull ? 0 : src_off[p + 1] - src_off[p];
        if (isnull && out_valid)
            atomicAnd((unsigned long long*)&out_valid[i >> 6], ~(1ull << (i & 63)));
    }
}

__global__ void k_gather_var_bytes(const uint8_t* __restrict__ src_bytes,
                                   const int32_t* __restrict__ src_off,
                                   const int32_t* __restrict__ pos, int32_t n,
                                   const int32_t* __restrict__ out_off,
                                   uint8_t* __restrict__ out_bytes)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int32_t len = out_off[i + 1] - out_off[i];
        if (pos[i] < 0) continue;   /* null: zero-length */
        const uint8_t* sp = src_bytes + src_off[pos[i]];
        uint8_t* dp = out_bytes + out_off[i];
        for (int32_t b = 0; b < len; b++) dp[b] = sp[b];
    }
}

static tg_status run_gather_var(tg_session* s, const DevBlock& src,
                                const int32_t* d_positions, int32_t count,
                                DevBlock* out)
{
    out->type = TG_VARCHAR;
    out->n = count;
    TG_POOL_ALLOC(s, &out->offsets, (int64_t)(count + 1) * 4);
    if (src.valid) {
        int64_t words = (count + 63) / 64;
        TG_POOL_ALLOC(s, &out->valid, (words ? words : 1) * 8);
        TG_HIP_CHECK(hipMemsetAsync(out->valid, 0xFF, words * 8, s->stream));
    }
    int grid = tg_grid_for(count ? count : 1);
    hipLaunchKernelGGL(k_gather_var_lens, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       src.offsets, src.valid, d_positions, count,
                       out->offsets, out->valid);
    TG_HIP_CHECK(hipGetLastError());
    int32_t* d_total = nullptr;
    TG_POOL_ALLOC(s, &d_total, 4);
    tg_status scst = run_scan_i32(s, out->offsets, count, d_total);
    if (scst != TG_OK) return scst;
    int32_t total = 0;
    TG_HIP_CHECK(hipMemcpyAsync(&total, d_total, 4, hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    TG_HIP_CHECK(hipMemcpyAsync(out->offsets + count, d_total, 4,
                                hipMemcpyDeviceToDevice, s->stream));
    TG_POOL_ALLOC(s, &out->data, total ? total : 1);
    hipLaunchKernelGGL(k_gather_var_bytes, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                       (const uint8_t*)src.data, src.offsets, d_positions, count,
                       out->offsets, (uint8_t*)out->data);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_total);
    return TG_OK;
}

tg_status run_gather(tg_session* s, const DevBlock& src, const int32_t* d_positions,
                     int32_t count, DevBlock* out, bool null_positions)
{
    if (src.type == TG_VARCHAR)
        return run_gather_var(s, src, d_positions, count, out);
    out->type = src.type;
    out->n = count;
    TG_POOL_ALLOC(s, &out->data, (int64_t)(count ? count : 1) * src.elem_size());
    if (src.valid || null_positions) {
        int64_t words = (count + 63) / 64;
        TG_POOL_ALLOC(s, &out->valid, (words ? words : 1) * 8);
        TG_HIP_CHECK(hipMemsetAsync(out->valid, 0xFF, words * 8, s->stream));
    }
    int grid = tg_grid_for(count);
    switch (src.elem_size()) {
        case 8: hipLaunchKernelGGL(k_gather<int64_t>, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                    (const int64_t*)src.data, src.valid, d_positions, count,
                    (int64_t*)out->data, out->valid); break;
        case 4: hipLaunchKernelGGL(k_gather<int32_t>, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                    (const int32_t*)src.data, src.valid, d_positions, count,
                    (int32_t*)out->data, out->valid); break;
        case 2: hipLaunchKernelGGL(k_gather<int16_t>, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                    (const int16_t*)src.data, src.valid, d_positions, count,
                    (int16_t*)out->data, out->valid); break;
        default: hipLaunchKernelGGL(k_gather<int8_t>, dim3(grid), dim3(TG_BLOCK), 0, s->stream,
                    (const int8_t*)src.data, src.valid, d_positions, count,
                    (int8_t*)out->data, out->valid); break;
    }
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

/* specialized projection a*(c-b) — the extendedprice*(1-discount) shape on
 * the Q3/Q6/Q14/Q19 paths; the postfix interpreter costs ~4x its memory
 * traffic on this (profiles/r01_q3_kernels_v3) */
__global__ void k_proj_mulsub(const double* __restrict__ a, const double* __restrict__ b,
                              double c, const int32_t* __restrict__ pos, int32_t n,
                              double* __restrict__ out)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int64_t p = pos ? (int64_t)pos[i] : i;   /* null = identity (no filter) */
        out[i] = a[p] * (c - b[p]);
    }
}

__global__ void k_proj_cmp_flag(KCol col, int op, double cval, int64_t icval,
                                int int_mode, const int32_t* __restrict__ pos,
                                int32_t n, double* __restrict__ out)
{
    int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; k < n; k += stride) {
        int64_t i = pos ? (int64_t)pos[k] : k;
        TVal v = load_tcol(col, i);
        bool r;
        if (int_mode) {
            switch (op) {
                case TG_EXPR_LE: r = v.i <= icval; break;
                case TG_EXPR_LT: r = v.i < icval; break;
                case TG_EXPR_GE: r = v.i >= icval; break;
                case TG_EXPR_GT: r = v.i > icval; break;
                case TG_EXPR_EQ: r = v.i == icval; break;
                default: r = v.i != icval; break;
            }
        }
        else {
            switch (op) {
                case TG_EXPR_LE: r = v.f <= cval; break;
                case TG_EXPR_LT: r = v.f < cval; break;
                case TG_EXPR_GE: r = v.f >= cval; break;
                case TG_EXPR_GT: r = v.f > cval; break;
                case TG_EXPR_EQ: r = v.f == cval; break;
                default: r = v.f != cval; break;
            }
        }
        out[k] = r ? 1.0 : 0.0;
    }
}

tg_status run_project(tg_session* s, const ExprProgram& proj, tg_type out_type,
                      const DevPage& page, const int32_t* d_positions, int32_t count,
                      DevBlock* out)
{
    /* identity projection = typed gather */
    if (proj.count == 1 && proj.insts[0].op == TG_EXPR_COL)
        return run_gather(s, page.blocks[proj.insts[0].arg0], d_positions, count, out);

    /* COL cmp CONST boolean projection (Q7/Q8's year flag): specialized
     * write of 1.0/0.0 — the interpreter cost ~10.7 ms over 180M rows */
    if (proj.count == 3 && proj.insts[0].op == TG_EXPR_COL &&
        (proj.insts[1].op == TG_EXPR_CONST_F64 || proj.insts[1].op == TG_EXPR_CONST_I64) &&
        proj.insts[2].op >= TG_EXPR_LE && proj.insts[2].op <= TG_EXPR_NE) {
        const DevBlock& a = page.blocks[proj.insts[0].arg0];
        if (!a.valid && a.type != TG_VARCHAR && out_type != TG_BIGINT) {
            bool is_i64c = proj.insts[1].op == TG_EXPR_CONST_I64;
            double cv = is_i64c ? (double)proj.insts[1].imm.i64
                                : proj.insts[1].imm.f64;
            int64_t icv = is_i64c ? proj.insts[1].imm.i64
                                  : (int64_t)proj.insts[1].imm.f64;
            KCol kc{a.data, nullptr, (int32_t)a.type, 0};
            out->type = TG_DOUBLE;
            out->n = count;
            TG_POOL_ALLOC(s, &out->data, (int64_t)(count ? count : 1) * 8);
            hipLaunchKernelGGL(k_proj_cmp_flag, dim3(tg_grid_for(count)),
                               dim3(TG_BLOCK), 0, s->stream, kc,
                               proj.insts[2].op, cv, icv,
                               is_i64c && type_is_int(a.type) ? 1 : 0,
                               d_positions, count, (double*)out->data);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            return TG_OK;
        }
    }

    /* a*(const-b) fast path (both DOUBLE, no nulls) */
    if (proj.count == 5 &&
        proj.insts[0].op == TG_EXPR_COL && proj.insts[1].op == TG_EXPR_CONST_F64 &&
        proj.insts[2].op == TG_EXPR_COL && proj.insts[3].op == TG_EXPR_SUB &&
        proj.insts[4].op == TG_EXPR_MUL) {
        const DevBlock& a = page.blocks[proj.insts[0].arg0];
        const DevBlock& b = page.blocks[proj.insts[2].arg0];
        if (a.type == TG_DOUBLE && b.type == TG_DOUBLE && !a.valid && !b.valid) {
            out->type = TG_DOUBLE;
            out->n = count;
            TG_POOL_ALLOC(s, &out->data, (int64_t)(count ? count : 1) * 8);
            hipLaunchKernelGGL(k_proj_mulsub, dim3(tg_grid_for(count)), dim3(TG_BLOCK),
                               0, s->stream, (const double*)a.data, (const double*)b.data,
                               proj.insts[1].imm.f64, d_positions, count, (double*)out->data);
            TG_HIP_CHECK(hipGetLastError());
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            return TG_OK;
        }
    }

    std::vector<KCol> cols(page.blocks.size());
    bool any_null = false;
    for (size_t i = 0; i < page.blocks.size(); i++) {
        cols[i] = {page.blocks[i].data, page.blocks[i].valid, (int32_t)page.blocks[i].type, 0};
        any_null |= page.blocks[i].valid != nullptr;
    }
    KCol* d_cols = nullptr;
    TG_POOL_ALLOC(s, &d_cols, cols.size() * sizeof(KCol));
    TG_HIP_CHECK(hipMemcpyAsync(d_cols, cols.data(), cols.size() * sizeof(KCol),
                                hipMemcpyHostToDevice, s->stream));
    int out_is_i64 = out_type == TG_BIGINT;
    out->type = out_is_i64 ? TG_BIGINT : TG_DOUBLE;
    out->n = count;
    TG_POOL_ALLOC(s, &out->data, (int64_t)(count ? count : 1) * 8);
    if (any_null) {
        int64_t words = (count + 63) / 64;
        TG_POOL_ALLOC(s, &out->valid, (words ? words : 1) * 8);
        TG_HIP_CHECK(hipMemsetAsync(out->valid, 0xFF, words * 8, s->stream));
    }
    hipLaunchKernelGGL(k_project, dim3(tg_grid_for(count)), dim3(TG_BLOCK), 0, s->stream,
                       proj.d_insts, proj.count, d_cols, d_positions, count,
                       out->data, out_is_i64, out->valid);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    tg_pool_free(s, d_cols);
    return TG_OK;
}

/* ---- FilterAndProject operator ---- */
bool tg_bridge_df(tg_join_bridge* b, const uint64_t** bm, int64_t* mn, int64_t* mx);

struct FilterProjectOp : tg_operator {
    ExprProgram filter{};
    bool has_filter = false;
    std::vector<ExprProgram> projections;
    std::vector<tg_type> out_types;
    tg_join_bridge* df_bridge = nullptr;    /* fused dynamic filter source */
    int32_t df_channel = -1;

    tg_status add_input(const tg_page* page) override
    {
        /* A/B on MI355X (Q3 SF100, same box, 2 runs each): fused two-pass
         * 128.4 ms/step vs selection-vector path 110.5 ms — the wave-per-chunk
         * write pass starves parallelism (37k waves doing serial 64-wide
         * evaluation) against grid-stride flag+gather kernels. Default stays
         * unfused; the fused path needs a block-parallel write design
         *  before it can win. */
        static int fused = [] { const char* e = getenv("TG_FP_FUSED"); return e ? atoi(e) : 0; }();
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        DF df{nullptr, 0, 0, 0, 0};
        bool have_df = false;
        if (df_bridge) {
            have_df = tg_bridge_df(df_bridge, &df.bm, &df.mn, &df.mx);
            df.col = df_channel;
            /* best effort (like the reference): no bitmap -> plain filter */
        }
        if ((has_filter || have_df) && !fused) {
            /* unfused reference path: selection vector + per-projection gather */
            int32_t* d_pos = nullptr;
            int32_t count = 0;
            st = run_filter(s, filter, in, nullptr, &d_pos, &count,
                            have_df ? &df : nullptr);
            if (st != TG_OK) { tg_free_page(s, &in); return st; }
            DevPage outp;
            outp.n = count;
            outp.blocks.resize(projections.size());
            for (size_t p = 0; p < projections.size(); p++) {
                st = run_project(s, projections[p], out_types[p], in, d_pos, count,
                                 &outp.blocks[p]);
                if (st != TG_OK) break;
            }
            if (d_pos) tg_pool_free(s, d_pos);
            tg_free_page(s, &in);
            if (st != TG_OK) return st;
            stage_output(std::move(outp));
            return TG_OK;
        }
        if (!has_filter) {
            /* pure projection: evaluate every position */
            DevPage outp;
            outp.n = in.n;
            outp.blocks.resize(projections.size());
            for (size_t p = 0; p < projections.size(); p++) {
                st = run_project(s, projections[p], out_types[p], in, nullptr,
                                 (int32_t)in.n, &outp.blocks[p]);
                if (st != TG_OK) { tg_free_page(s, &in); return st; }
            }
            tg_free_page(s, &in);
            stage_output(std::move(outp));
            return TG_OK;
        }
        if (projections.size() > MAX_PROJ) { TG_SET_ERR("too many projections"); return TG_ERR_UNSUPPORTED; }

        std::vector<KCol> cols(in.blocks.size());
        for (size_t i = 0; i < in.blocks.size(); i++)
            cols[i] = {in.blocks[i].data, in.blocks[i].valid, (int32_t)in.blocks[i].type, 0};
        KCol* d_cols = nullptr;
        TG_POOL_ALLOC(s, &d_cols, cols.size() * sizeof(KCol));
        TG_HIP_CHECK(hipMemcpyAsync(d_cols, cols.data(), cols.size() * sizeof(KCol),
                                    hipMemcpyHostToDevice, s->stream));

        int64_t nchunks = (in.n + CHUNK - 1) / CHUNK;
        if (nchunks < 1) nchunks = 1;
        int32_t* d_counts = nullptr;
        int32_t* d_total = nullptr;
        TG_POOL_ALLOC(s, &d_counts, nchunks * 4);
        TG_POOL_ALLOC(s, &d_total, 4);
        hipLaunchKernelGGL(k_fp_count, dim3((uint32_t)nchunks), dim3(TG_BLOCK), 0, s->stream,
                           filter.d_insts, filter.count, d_cols, in.n, d_counts, nchunks);
        TG_HIP_CHECK(hipGetLastError());
        tg_status fsst = run_scan_i32(s, d_counts, nchunks, d_total);
        if (fsst != TG_OK) return fsst;
        int32_t total = 0;
        TG_HIP_CHECK(hipMemcpyAsync(&total, d_total, 4, hipMemcpyDeviceToHost, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));

        DevPage outp;
        outp.n = total;
        outp.blocks.resize(projections.size());
        std::vector<KProj> kp(projections.size());
        bool any_null = false;
        for (auto& b : in.blocks) any_null |= b.valid != nullptr;
        for (size_t p = 0; p < projections.size(); p++) {
            DevBlock& ob = outp.blocks[p];
            bool ident = projections[p].count == 1 &&
                         projections[p].insts[0].op == TG_EXPR_COL;
            int src = ident ? projections[p].insts[0].arg0 : -1;
            ob.type = ident ? in.blocks[src].type
                            : (out_types[p] == TG_BIGINT ? TG_BIGINT : TG_DOUBLE);
            ob.n = total;
            TG_POOL_ALLOC(s, &ob.data, (int64_t)(total ? total : 1) * ob.elem_size());
            if (any_null) {
                int64_t words = (total + 63) / 64;
                TG_POOL_ALLOC(s, &ob.valid, (words ? words : 1) * 8);
                TG_HIP_CHECK(hipMemsetAsync(ob.valid, 0xFF, words * 8, s->stream));
            }
            kp[p] = {projections[p].d_insts, projections[p].count, src,
                     (int32_t)ob.type, ob.data, ob.valid};
        }
        KProj* d_kp = nullptr;
        TG_POOL_ALLOC(s, &d_kp, kp.size() * sizeof(KProj));
        TG_HIP_CHECK(hipMemcpyAsync(d_kp, kp.data(), kp.size() * sizeof(KProj),
                                    hipMemcpyHostToDevice, s->stream));
        hipLaunchKernelGGL(k_fp_write, dim3((uint32_t)nchunks), dim3(TG_BLOCK), 0, s->stream,
                           filter.d_insts, filter.count, d_cols, in.n, d_counts,
                           d_kp, (int)kp.size(), nchunks);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_cols);
        tg_pool_free(s, d_counts);
        tg_pool_free(s, d_total);
        tg_pool_free(s, d_kp);
        tg_free_page(s, &in);
        stage_output(std::move(outp));
        return TG_OK;
    }

    tg_status get_output(tg_page* out, int* finished) override
    {
        emit_staged(out, finished);
        return TG_OK;
    }

    ~FilterProjectOp() override
    {
        tg_free_expr(s, &filter);
        for (auto& p : projections) tg_free_expr(s, &p);
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

static tg_status check_expr_depth(const tg_expr* e)
{
    int sp = 0, maxsp = 0;
    for (int i = 0; i < e->count; i++) {
        switch (e->insts[i].op) {
            case TG_EXPR_COL: case TG_EXPR_CONST_F64: case TG_EXPR_CONST_I64: sp++; break;
            case TG_EXPR_NOT: break;
            case TG_EXPR_BETWEEN: sp -= 2; break;
            default: sp -= 1; break;
        }
        if (sp > maxsp) maxsp = sp;
        if (sp < 1) { TG_SET_ERR("expr stack underflow at inst %d", i); return TG_ERR_INVALID_ARG; }
    }
    if (maxsp > MAX_STACK) { TG_SET_ERR("expr stack depth %d > %d", maxsp, MAX_STACK); return TG_ERR_UNSUPPORTED; }
    if (sp != 1) { TG_SET_ERR("expr does not reduce to one value"); return TG_ERR_INVALID_ARG; }
    return TG_OK;
}

extern "C" tg_status tg_filter_project_create_df(tg_session* s,
    const tg_expr* filter, const tg_expr* projections,
    const int32_t* proj_out_types, int32_t n_proj, tg_join_bridge* df_bridge,
    int32_t df_key_channel, tg_operator** out)
{
    tg_status st = tg_filter_project_create(s, filter, projections,
                                            proj_out_types, n_proj, out);
    if (st != TG_OK) return st;
    auto* op = static_cast<FilterProjectOp*>(*out);
    op->df_bridge = df_bridge;
    op->df_channel = df_key_channel;
    return TG_OK;
}

extern "C" tg_status tg_filter_project_create(tg_session* s, const tg_expr* filter,
    const tg_expr* projections, const int32_t* proj_out_types, int32_t n_proj,
    tg_operator** out)
{
    if (!s || !out || (n_proj > 0 && !projections)) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    auto* op = new FilterProjectOp();
    op->s = s;
    tg_status st = TG_OK;
    if (filter) {
        st = check_expr_depth(filter);
        if (st == TG_OK) st = tg_compile_expr(s, filter, &op->filter);
        op->has_filter = true;
    }
    for (int p = 0; st == TG_OK && p < n_proj; p++) {
        tg_expr e{projections[p].insts, projections[p].count};
        st = check_expr_depth(&e);
        if (st != TG_OK) break;
        op->projections.emplace_back();
        st = tg_compile_expr(s, &e, &op->projections.back());
        op->out_types.push_back(proj_out_types ? (tg_type)proj_out_types[p] : TG_DOUBLE);
    }
    if (st != TG_OK) { delete op; return st; }
    *out = op;
    return TG_OK;
}


/* ---- dictionary-aware filter (sql/gen/columnar/
 * DictionaryAwareColumnarFilter.java:44-80): when every column reference in
 * the predicate is one dictionary-encoded channel, evaluate the predicate
 * ONCE PER DICTIONARY ENTRY, then each row tests verdict[id] — 4 B/row of
 * ids instead of decoding + re-reading full values. ---- */
__global__ void k_expand_dict_flags(const uint8_t* __restrict__ dflags,
                                    const int32_t* __restrict__ ids, int64_t n,
                                    uint8_t* __restrict__ flags)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) flags[i] = dflags[ids[i]];
}

static bool expr_single_col(const tg_expr* e, int32_t* col)
{
    int32_t c = -1;
    for (int i = 0; i < e->count; i++) {
        if (e->insts[i].op == TG_EXPR_COL) {
            if (c >= 0 && e->insts[i].arg0 != c) return false;
            c = e->insts[i].arg0;
        }
    }
    *col = c;
    return c >= 0;
}

static tg_status run_filter_dict(tg_session* s, const ExprProgram& pred,
                                 const tg_block* b /* dictionary block */,
                                 int32_t col_in_expr,
                                 int32_t** d_positions_out, int32_t* count_out)
{
    const tg_block* dict = b->dictionary;
    int64_t nd = dict->position_count;
    int64_t n = b->position_count;
    /* upload dictionary values as a single flat column */
    size_t esz = 8;
    switch ((tg_type)dict->type) {
        case TG_INTEGER: case TG_DATE: esz = 4; break;
        case TG_SMALLINT: esz = 2; break;
        case TG_TINYINT: case TG_BOOLEAN: esz = 1; break;
        default: esz = 8; break;
    }
    void* d_dv = nullptr;
    tg_status st = upload_flat(s, dict->data, dict->on_device, nd * esz, &d_dv);
    if (st != TG_OK) return st;
    /* KCol table sized to the expr's channel index (others unused) */
    std::vector<KCol> cols((size_t)col_in_expr + 1);
    cols[col_in_expr] = {d_dv, nullptr, dict->type, 0};
    KCol* d_cols = nullptr;
    TG_POOL_ALLOC(s, &d_cols, cols.size() * sizeof(KCol));
    TG_HIP_CHECK(hipMemcpyAsync(d_cols, cols.data(), cols.size() * sizeof(KCol),
                                hipMemcpyHostToDevice, s->stream));
    uint8_t* d_dflags = nullptr;
    TG_POOL_ALLOC(s, &d_dflags, nd ? nd : 1);
    DF nodf{nullptr, 0, 0, 0, 0};
    hipLaunchKernelGGL(k_filter_flags, dim3(tg_grid_for(nd)), dim3(TG_BLOCK), 0,
                       s->stream, pred.d_insts, pred.count, d_cols, 0, nullptr,
                       0, nd, d_dflags, nodf);
    TG_HIP_CHECK(hipGetLastError());
    /* expand verdicts over the ids */
    int32_t* d_ids = nullptr;
    st = upload_flat(s, b->ids, b->on_device, n * 4, (void**)&d_ids);
    if (st != TG_OK) return st;
    uint8_t* d_flags = nullptr;
    TG_POOL_ALLOC(s, &d_flags, n ? n : 1);
    hipLaunchKernelGGL(k_expand_dict_flags, dim3(tg_grid_for(n)), dim3(TG_BLOCK),
                       0, s->stream, d_dflags, d_ids, n, d_flags);
    TG_HIP_CHECK(hipGetLastError());
    /* count + compact (same shape as run_filter's tail) */
    int64_t nchunks = (n + CHUNK - 1) / CHUNK;
    if (nchunks < 1) nchunks = 1;
    int32_t* d_offsets = nullptr;
    int32_t* d_total = nullptr;
    TG_POOL_ALLOC(s, &d_offsets, nchunks * 4);
    TG_POOL_ALLOC(s, &d_total, 4);
    hipLaunchKernelGGL(k_count_chunk,
                       dim3((uint32_t)((nchunks + TG_BLOCK / 64 - 1) / (TG_BLOCK / 64))),
                       dim3(TG_BLOCK), 0, s->stream, d_flags, n, d_offsets, nchunks);
    TG_HIP_CHECK(hipGetLastError());
    st = run_scan_i32(s, d_offsets, nchunks, d_total);
    if (st != TG_OK) return st;
    int32_t total = 0;
    TG_HIP_CHECK(hipMemcpyAsync(&total, d_total, 4, hipMemcpyDeviceToHost, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    int32_t* d_pos = nullptr;
    TG_POOL_ALLOC(s, &d_pos, (total ? total : 1) * 4);
    hipLaunchKernelGGL(k_compact, dim3((int)nchunks), dim3(64), 0, s->stream,
                       d_flags, n, d_offsets, 0, nullptr, 0, d_pos);
    TG_HIP_CHECK(hipGetLastError());
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    for (void* q : {(void*)d_dv, (void*)d_cols, (void*)d_dflags, (void*)d_ids,
                    (void*)d_flags, (void*)d_offsets, (void*)d_total})
        tg_pool_free(s, q);
    *d_positions_out = d_pos;
    *count_out = total;
    return TG_OK;
}

extern "C" tg_status tg_filter_run(tg_session* s, const tg_expr* filter, const tg_page* page,
                                   const tg_selected* input_sel,
                                   int32_t* out_positions, int32_t* out_count)
{
    if (!s || !filter || !page || !out_positions || !out_count) {
        TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG;
    }
    tg_status st = check_expr_depth(filter);
    if (st != TG_OK) return st;
    ExprProgram prog;
    st = tg_compile_expr(s, filter, &prog);
    if (st != TG_OK) return st;
    /* dictionary-aware path: single-channel predicate over a fixed-width
     * dictionary block, and no input selection (whole-page evaluation) */
    int32_t fc = -1;
    if (!input_sel && expr_single_col(filter, &fc) &&
        fc < page->channel_count &&
        page->blocks[fc].kind == TG_BK_DICTIONARY &&
        page->blocks[fc].dictionary &&
        page->blocks[fc].dictionary->type != TG_VARCHAR &&
        !page->blocks[fc].dictionary->valid) {
        int32_t* d_pos = nullptr;
        int32_t count = 0;
        st = run_filter_dict(s, prog, &page->blocks[fc], fc, &d_pos, &count);
        if (st == TG_OK) {
            TG_HIP_CHECK(hipMemcpyAsync(out_positions, d_pos, count * 4,
                                        hipMemcpyDeviceToHost, s->stream));
            TG_HIP_CHECK(hipStreamSynchronize(s->stream));
            *out_count = count;
        }
        if (d_pos) tg_pool_free(s, d_pos);
        tg_free_expr(s, &prog);
        return st;
    }
    DevPage in;
    st = tg_upload_page(s, page, &in);
    if (st != TG_OK) { tg_free_expr(s, &prog); return st; }
    int32_t* d_pos = nullptr;
    int32_t count = 0;
    st = run_filter(s, prog, in, input_sel, &d_pos, &count);
    if (st == TG_OK) {
        TG_HIP_CHECK(hipMemcpyAsync(out_positions, d_pos, count * sizeof(int32_t),
                                    hipMemcpyDeviceToHost, s->stream));
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        *out_count = count;
    }
    if (d_pos) tg_pool_free(s, d_pos);
    tg_free_page(s, &in);
    tg_free_expr(s, &prog);
    return st;
}
