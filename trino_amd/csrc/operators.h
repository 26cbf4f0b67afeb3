/* operators.h — internal operator framework of libtrino_gpu.
 * Mirrors operator/Operator.java:18-50 state machine: addInput/getOutput/
 * finish, driven single-threaded per operator (Driver.processInternal,
 * operator/Driver.java:391-425). Host C++; all data device-resident.
 */
#pragma once
#include "common.h"
#include <vector>
#include <memory>

/* device-resident column (flat Block mirror) */
struct DevBlock {
    tg_type type = TG_BIGINT;
    int64_t n = 0;
    void* data = nullptr;          /* owned device buffer */
    uint64_t* valid = nullptr;     /* packed bitmap, bit=1 valid; null = no nulls */
    int32_t* offsets = nullptr;    /* VARCHAR: n+1 offsets into data */
    bool owned = true;
    bool valid_owned_override = false;   /* data owned but bitmap borrowed */
    int64_t elem_size() const
    {
        switch (type) {
            case TG_BIGINT: case TG_DOUBLE: return 8;
            case TG_INTEGER: case TG_DATE: return 4;
            case TG_SMALLINT: return 2;
            default: return 1;
        }
    }
};

struct DevPage {
    int64_t n = 0;
    std::vector<DevBlock> blocks;
};

/* upload a host/device tg_page into device-owned DevPage (flat memcpy of the
 * backing arrays; DictionaryBlock/RLE are decoded to flat values on upload —
 * the aggregation loops only need the flat ValueBlock forms, SURVEY.md §8
 * micro-semantics). */
tg_status tg_upload_page(tg_session* s, const tg_page* in, DevPage* out);
void tg_free_page(tg_session* s, DevPage* p);

struct tg_operator {
    tg_session* s = nullptr;
    bool input_finished = false;
    virtual int needs_input() { return !input_finished; }
    virtual tg_status add_input(const tg_page* page) = 0;
    virtual tg_status get_output(tg_page* out, int* finished) = 0;
    virtual tg_status finish() { input_finished = true; return TG_OK; }
    virtual ~tg_operator() = default;

    /* storage backing the tg_page returned by get_output (valid until next call) */
    std::vector<tg_block> out_blocks_;
    std::vector<DevPage> out_pages_;
    size_t next_out_ = 0;

    void stage_output(DevPage&& p) { out_pages_.emplace_back(std::move(p)); }
    bool emit_staged(tg_page* out, int* finished)
    {
        if (next_out_ < out_pages_.size()) {
            DevPage& p = out_pages_[next_out_++];
            out_blocks_.clear();
            for (auto& b : p.blocks) {
                tg_block tb{};
                tb.type = b.type;
                tb.kind = TG_BK_VALUE;
                tb.position_count = p.n;
                tb.on_device = 1;
                tb.data = b.data;
                tb.valid = b.valid;
                tb.offsets = b.offsets;
                out_blocks_.push_back(tb);
            }
            out->channel_count = (int32_t)p.blocks.size();
            out->position_count = p.n;
            out->blocks = out_blocks_.data();
            *finished = 0;
            return true;
        }
        out->channel_count = 0;
        out->position_count = 0;
        out->blocks = nullptr;
        *finished = input_finished ? 1 : 0;
        return false;
    }
};

/* expression IR (host-compiled to a device-constant program) */
struct ExprProgram {
    std::vector<tg_expr_inst> insts;
    tg_expr_inst* d_insts = nullptr;   /* device copy */
    int count = 0;
};
tg_status tg_compile_expr(tg_session* s, const tg_expr* e, ExprProgram* out);
void tg_free_expr(tg_session* s, ExprProgram* p);

/* kernels (implemented in ops_*.hip) */
struct DF;   /* fused dynamic filter spec (ops_filter.hip) */
tg_status upload_flat(tg_session* s, const void* src, int on_device,
                      int64_t bytes, void** out);
tg_status run_filter(tg_session* s, const ExprProgram& pred, const DevPage& page,
                     const tg_selected* input_sel,
                     int32_t** d_positions_out, int32_t* count_out,
                     const DF* df = nullptr);
tg_status run_project(tg_session* s, const ExprProgram& proj, tg_type out_type,
                      const DevPage& page, const int32_t* d_positions, int32_t count,
                      DevBlock* out);
tg_status run_gather(tg_session* s, const DevBlock& src, const int32_t* d_positions,
                     int32_t count, DevBlock* out, bool null_positions = false);
tg_status run_hash_rows(tg_session* s, const DevPage& page,
                        const int32_t* channels, int32_t n_channels,
                        uint64_t* d_hashes);

/* device radix-sort helpers (sort.hip; rocPRIM-backed, stable) */
tg_status run_sort_pairs(tg_session* s, uint64_t* d_keys, int64_t* d_vals, int64_t n);
tg_status run_argsort_i64(tg_session* s, int64_t* d_keys, int64_t n, int32_t* d_out_idx);
tg_status run_sort_keys_u32(tg_session* s, uint32_t* d_keys, int64_t n);
tg_status run_sort_pairs_bits(tg_session* s, uint64_t* d_keys, int64_t* d_vals,
                              int64_t n, int bits);
tg_status run_sort_pairs_u32(tg_session* s, uint32_t* d_keys, int32_t* d_vals, int64_t n);
tg_status run_scan_i32(tg_session* s, int32_t* d_arr, int64_t n, int32_t* d_total);
