/* ops_topn.hip — TopN operator (§8f row 2).
 *
 * Mirrors operator/TopNOperator.java / TopNProcessor: accumulate input pages,
 * keep the top `limit` rows under the given orderings, emit one page in order.
 * Orderings follow Trino's defaults: ASC = nulls last, DESC = nulls first.
 *
 * Round-1 shape: input pages stay device-resident; at finish the SORT-KEY
 * channels of the accumulated rows are brought to the host, a partial_sort
 * selects the top-limit indices (limit is small — Q3/most sweep tails use
 * LIMIT 10..100), and the output page is gathered on device. A device top-k
 * selection kernel replaces the host sort when limit*pages grows (tracked in
 * DESIGN.md §8f).
 */
#include "dev_hash.h"
#include <algorithm>

struct TopNOp : tg_operator {
    std::vector<tg_type> types;
    std::vector<int32_t> sort_channels;
    std::vector<int32_t> sort_desc;     /* 0 asc, 1 desc */
    int32_t limit = 0;
    std::vector<DevPage> pages;
    int64_t total_rows = 0;
    bool emitted = false;

    tg_status add_input(const tg_page* page) override
    {
        DevPage in;
        tg_status st = tg_upload_page(s, page, &in);
        if (st != TG_OK) return st;
        total_rows += in.n;
        pages.emplace_back(std::move(in));
        return TG_OK;
    }

    tg_status emit()
    {
        /* concatenate sort channels on host */
        size_t k = sort_channels.size();
        std::vector<std::vector<double>> keys_f(k);
        std::vector<std::vector<int64_t>> keys_i(k);
        std::vector<std::vector<uint8_t>> keys_null(k);
        for (size_t j = 0; j < k; j++) {
            int ch = sort_channels[j];
            bool isf = types[ch] == TG_DOUBLE;
            if (isf) keys_f[j].reserve(total_rows); else keys_i[j].reserve(total_rows);
            keys_null[j].reserve(total_rows);
            for (auto& p : pages) {
                const DevBlock& b = p.blocks[ch];
                std::vector<uint64_t> valid;
                if (b.valid) {
                    valid.resize((p.n + 63) / 64);
                    TG_HIP_CHECK(hipMemcpy(valid.data(), b.valid, valid.size() * 8,
                                           hipMemcpyDeviceToHost));
                }
                if (isf) {
                    std::vector<double> tmp(p.n);
                    TG_HIP_CHECK(hipMemcpy(tmp.data(), b.data, p.n * 8, hipMemcpyDeviceToHost));
                    keys_f[j].insert(keys_f[j].end(), tmp.begin(), tmp.end());
                }
                else {
                    std::vector<int64_t> tmp(p.n);
                    switch (b.elem_size()) {
                        case 8: {
                            TG_HIP_CHECK(hipMemcpy(tmp.data(), b.data, p.n * 8, hipMemcpyDeviceToHost));
                            break;
                        }
                        case 4: {
                            std::vector<int32_t> t4(p.n);
                            TG_HIP_CHECK(hipMemcpy(t4.data(), b.data, p.n * 4, hipMemcpyDeviceToHost));
                            for (int64_t i = 0; i < p.n; i++) tmp[i] = t4[i];
                            break;
                        }
                        case 2: {
                            std::vector<int16_t> t2(p.n);
                            TG_HIP_CHECK(hipMemcpy(t2.data(), b.data, p.n * 2, hipMemcpyDeviceToHost));
                            for (int64_t i = 0; i < p.n; i++) tmp[i] = t2[i];
                            break;
                        }
                        default: {
                            std::vector<int8_t> t1(p.n);
                            TG_HIP_CHECK(hipMemcpy(t1.data(), b.data, p.n, hipMemcpyDeviceToHost));
                            for (int64_t i = 0; i < p.n; i++) tmp[i] = t1[i];
                            break;
                        }
                    }
                    keys_i[j].insert(keys_i[j].end(), tmp.begin(), tmp.end());
                }
                for (int64_t i = 0; i < p.n; i++) {
                    bool isnull = b.valid && !((valid[i >> 6] >> (i & 63)) & 1);
                    keys_null[j].push_back(isnull ? 1 : 0);
                }
            }
        }
        /* comparator: channel-ordered; ASC nulls last, DESC nulls first */
        auto less = [&](int64_t a, int64_t b) {
            for (size_t j = 0; j < k; j++) {
                bool desc = sort_desc[j];
                bool na = keys_null[j][a], nb = keys_null[j][b];
                if (na || nb) {
                    if (na != nb) return desc ? na : nb;  /* null first on desc */
                    continue;
                }
                int cmp;
                if (types[sort_channels[j]] == TG_DOUBLE) {
                    double x = keys_f[j][a], y = keys_f[j][b];
                    cmp = (x < y) ? -1 : (x > y) ? 1 : 0;
                }
                else {
                    int64_t x = keys_i[j][a], y = keys_i[j][b];
                    cmp = (x < y) ? -1 : (x > y) ? 1 : 0;
                }
                if (cmp) return desc ? cmp > 0 : cmp < 0;
            }
            return false;
        };
        int64_t take = std::min<int64_t>(limit, total_rows);
        std::vector<int64_t> idx(total_rows);
        for (int64_t i = 0; i < total_rows; i++) idx[i] = i;
        std::partial_sort(idx.begin(), idx.begin() + take, idx.end(), less);
        idx.resize(take);

        /* map flat row -> (page, row) and gather per page (order preserved by
         * gathering per output slot via int32 positions within each page) */
        std::vector<int64_t> page_base(pages.size() + 1, 0);
        for (size_t p = 0; p < pages.size(); p++)
            page_base[p + 1] = page_base[p] + pages[p].n;

        DevPage outp;
        outp.n = take;
        for (size_t c = 0; c < types.size(); c++) {
            DevBlock b;
            b.type = types[c];
            b.n = take;
            TG_POOL_ALLOC(s, &b.data, (take ? take : 1) * b.elem_size());
            /* copy element by element via DtoD (take is small: <= limit) */
            for (int64_t o = 0; o < take; o++) {
                int64_t flat = idx[o];
                size_t p = std::upper_bound(page_base.begin(), page_base.end(), flat) -
                           page_base.begin() - 1;
                int64_t row = flat - page_base[p];
                TG_HIP_CHECK(hipMemcpyAsync((char*)b.data + o * b.elem_size(),
                                            (char*)pages[p].blocks[c].data + row * b.elem_size(),
                                            b.elem_size(), hipMemcpyDeviceToDevice, s->stream));
            }
            outp.blocks.push_back(b);
        }
        TG_HIP_CHECK(hipStreamSynchronize(s->stream));
        stage_output(std::move(outp));
        for (auto& p : pages) tg_free_page(s, &p);
        pages.clear();
        return TG_OK;
    }

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

    ~TopNOp() override
    {
        for (auto& p : pages) tg_free_page(s, &p);
        for (auto& p : out_pages_) tg_free_page(s, &p);
    }
};

extern "C" tg_status tg_topn_create(tg_session* s,
    const int32_t* types, int32_t n_channels,
    const int32_t* sort_channels, const int32_t* sort_desc, int32_t n_sort,
    int32_t limit, tg_operator** out)
{
    if (!s || !types || !sort_channels || !sort_desc || limit < 1 || n_sort < 1) {
        TG_SET_ERR("invalid topn spec");
        return TG_ERR_INVALID_ARG;
    }
    auto* op = new TopNOp();
    op->s = s;
    for (int i = 0; i < n_channels; i++) op->types.push_back((tg_type)types[i]);
    op->sort_channels.assign(sort_channels, sort_channels + n_sort);
    op->sort_desc.assign(sort_desc, sort_desc + n_sort);
    op->limit = limit;
    *out = op;
    return TG_OK;
}
