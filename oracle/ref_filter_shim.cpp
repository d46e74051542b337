/* ORACLE — test infrastructure only. Compiles the REFERENCE's own SQL
 * residual-filter evaluator (cgo/cuvs/filter.hpp — header-only, std+omp)
 * behind a flat C ABI so tests/test_preds.py can pin matrixone_amd's
 * independent moann_preds.cpp implementation bit-for-bit against the
 * reference's eval_filter_bitmap_cpu on identical inputs. Built by
 * `make -C oracle filters_ref` when /root/reference exists; never linked
 * or loaded by the product path. */

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

#include "filter.hpp"

extern "C" int ref_preds_eval(const char* cols_json, const char* preds_json,
                              const void* const* col_data,
                              const uint32_t* const* col_nulls,
                              const int64_t* host_ids, uint64_t nrows,
                              uint32_t* out_words) {
    try {
        auto cols = matrixone::parse_filter_col_meta(
            cols_json ? cols_json : "");
        matrixone::FilterStore fs;
        fs.init(cols, nrows);
        for (size_t c = 0; c < cols.size(); ++c)
            fs.add_chunk((uint32_t)c, col_data[c],
                         col_nulls ? col_nulls[c] : nullptr, nrows);
        matrixone::HostIdsView hv;
        if (host_ids) {
            hv.data = host_ids;
            hv.type = matrixone::FilterColType::INT64;
            hv.count = nrows;
        }
        auto preds = matrixone::parse_preds(preds_json ? preds_json : "");
        auto mask = matrixone::eval_filter_bitmap_cpu(fs, preds, 0, nrows, hv);
        std::memcpy(out_words, mask.data(), mask.size() * 4);
        return 0;
    } catch (...) {
        return -1;
    }
}
