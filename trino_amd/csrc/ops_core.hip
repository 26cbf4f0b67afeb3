/* ops_core.hip — page upload/decode, expression compile, generic operator
 * C-ABI glue (operator/Operator.java:18-50 surface). */
#include "operators.h"

/* ---- dictionary / RLE decode kernels (collapse to flat values on upload;
 * DictionaryBlock.java:55-58, RunLengthEncodedBlock) ---- */
template <typename T>
__global__ void k_dict_decode(const T* __restrict__ dict, const int32_t* __restrict__ ids,
                              int64_t n, T* __restrict__ out)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = dict[ids[i]];
}

template <typename T>
__global__ void k_rle_decode(const T* __restrict__ value, int64_t n, T* __restrict__ out)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    T v = *value;
    for (; i < n; i += stride) out[i] = v;
}

tg_status upload_flat(tg_session* s, const void* src, int on_device,
                      int64_t bytes, void** out)
{
    TG_POOL_ALLOC(s, out, bytes);
    TG_HIP_CHECK(hipMemcpyAsync(*out, src, bytes,
                                on_device ? hipMemcpyDeviceToDevice : hipMemcpyHostToDevice,
                                s->stream));
    return TG_OK;
}

template <typename T>
static tg_status decode_to_flat(tg_session* s, const tg_block* b, DevBlock* out)
{
    int64_t n = b->position_count;
    TG_POOL_ALLOC(s, &out->data, n * sizeof(T));
    if (b->kind == TG_BK_DICTIONARY) {
        const tg_block* d = b->dictionary;
        void* d_dict = nullptr; int32_t* d_ids = nullptr;
        tg_status st = upload_flat(s, d->data, d->on_device, d->position_count * sizeof(T), &d_dict);
        if (st != TG_OK) return st;
        st = upload_flat(s, b->ids, b->on_device, n * sizeof(int32_t), (void**)&d_ids);
        if (st != TG_OK) return st;
        hipLaunchKernelGGL(k_dict_decode<T>, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                           (const T*)d_dict, d_ids, n, (T*)out->data);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_dict);
        tg_pool_free(s, d_ids);
    }
    else { /* RLE */
        const tg_block* d = b->dictionary;
        void* d_val = nullptr;
        tg_status st = upload_flat(s, d->data, d->on_device, sizeof(T), &d_val);
        if (st != TG_OK) return st;
        hipLaunchKernelGGL(k_rle_decode<T>, dim3(tg_grid_for(n)), dim3(TG_BLOCK), 0, s->stream,
                           (const T*)d_val, n, (T*)out->data);
        TG_HIP_CHECK(hipGetLastError());
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        tg_pool_free(s, d_val);
    }
    return TG_OK;
}

tg_status tg_upload_page(tg_session* s, const tg_page* in, DevPage* out)
{
    out->n = in->position_count;
    out->blocks.resize(in->channel_count);
    for (int c = 0; c < in->channel_count; c++) {
        const tg_block* b = &in->blocks[c];
        DevBlock* db = &out->blocks[c];
        db->type = (tg_type)b->type;
        db->n = b->position_count;
        if (b->type == TG_VARCHAR) {
            if (b->kind != TG_BK_VALUE || !b->offsets) {
                TG_SET_ERR("VARCHAR blocks cross as flat offsets+bytes or "
                           "dictionary-encoded ids");
                return TG_ERR_UNSUPPORTED;
            }
            /* VariableWidthBlock: int32 offsets[n+1] + utf8 bytes */
            int32_t total_bytes = 0;
            if (b->on_device) {
                TG_HIP_CHECK(hipMemcpy(&total_bytes, b->offsets + db->n, 4,
                                       hipMemcpyDeviceToHost));
            }
            else {
                total_bytes = b->offsets[db->n];
            }
            tg_status st = upload_flat(s, b->offsets, b->on_device,
                                       (db->n + 1) * 4, (void**)&db->offsets);
            if (st != TG_OK) return st;
            st = upload_flat(s, b->data, b->on_device,
                             total_bytes ? total_bytes : 1, &db->data);
            if (st != TG_OK) return st;
            if (b->valid) {
                int64_t words = (db->n + 63) / 64;
                st = upload_flat(s, b->valid, b->on_device, words * 8,
                                 (void**)&db->valid);
                if (st != TG_OK) return st;
            }
            continue;
        }
        if (b->kind == TG_BK_VALUE) {
            if (b->on_device) {
                /* zero-copy borrow: device-resident value blocks flow between
                 * operators without a DtoD pass (the producer keeps them
                 * alive until its close — Operator output contract). */
                db->data = const_cast<void*>(b->data);
                db->owned = false;
            }
            else {
                tg_status st = upload_flat(s, b->data, b->on_device,
                                           db->n * db->elem_size(), &db->data);
                if (st != TG_OK) return st;
            }
        }
        else {
            tg_status st;
            switch (db->elem_size()) {
                case 8: st = decode_to_flat<int64_t>(s, b, db); break;
                case 4: st = decode_to_flat<int32_t>(s, b, db); break;
                case 2: st = decode_to_flat<int16_t>(s, b, db); break;
                default: st = decode_to_flat<int8_t>(s, b, db); break;
            }
            if (st != TG_OK) return st;
        }
        if (b->valid) {
            if (b->on_device) {
                db->valid = const_cast<uint64_t*>(b->valid);
                /* owned already false for borrowed VALUE data; for decoded
                 * dictionary/RLE blocks the bitmap is borrowed separately */
                if (db->owned) db->valid_owned_override = true;
            }
            else {
                int64_t words = (db->n + 63) / 64;
                tg_status st = upload_flat(s, b->valid, b->on_device, words * 8,
                                           (void**)&db->valid);
                if (st != TG_OK) return st;
            }
        }
    }
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

void tg_free_page(tg_session* s, DevPage* p)
{
    (void)s;
    for (auto& b : p->blocks) {
        if (b.owned && b.data) tg_pool_free(s, b.data);
        if (b.owned && !b.valid_owned_override && b.valid) tg_pool_free(s, b.valid);
        if (b.owned && b.offsets) tg_pool_free(s, b.offsets);
    }
    p->blocks.clear();
    p->n = 0;
}

tg_status tg_compile_expr(tg_session* s, const tg_expr* e, ExprProgram* out)
{
    out->insts.assign(e->insts, e->insts + e->count);
    out->count = e->count;
    if (out->count > 128) { TG_SET_ERR("expression too long (>128 insts)"); return TG_ERR_UNSUPPORTED; }
    TG_POOL_ALLOC(s, &out->d_insts, out->count * sizeof(tg_expr_inst));
    TG_HIP_CHECK(hipMemcpyAsync(out->d_insts, out->insts.data(),
                                out->count * sizeof(tg_expr_inst),
                                hipMemcpyHostToDevice, s->stream));
    TG_HIP_CHECK(hipStreamSynchronize(s->stream));
    return TG_OK;
}

void tg_free_expr(tg_session* s, ExprProgram* p)
{
    if (p->d_insts) tg_pool_free(s, p->d_insts);
    p->d_insts = nullptr;
}

/* ---- generic operator C ABI ---- */
extern "C" int tg_operator_needs_input(tg_operator* op)
{
    return op ? op->needs_input() : 0;
}

extern "C" tg_status tg_operator_add_input(tg_operator* op, const tg_page* page)
{
    if (!op || !page) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    if (op->input_finished) { TG_SET_ERR("addInput after finish"); return TG_ERR_STATE; }
    return op->add_input(page);
}

extern "C" tg_status tg_operator_get_output(tg_operator* op, tg_page* out, int* finished)
{
    if (!op || !out || !finished) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    return op->get_output(out, finished);
}

extern "C" tg_status tg_operator_finish(tg_operator* op)
{
    if (!op) { TG_SET_ERR("null arg"); return TG_ERR_INVALID_ARG; }
    return op->finish();
}

extern "C" void tg_operator_close(tg_operator* op)
{
    delete op;
}
