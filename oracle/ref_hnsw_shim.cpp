/* ORACLE / _ref tooling — test infrastructure ONLY.
 *
 * Drives the reference's own HNSW implementation (usearch 2.25.3, vendored at
 * /root/reference/thirdparties, the engine behind pkg/vectorindex/hnsw —
 * model.go:881-901 SearchUnsafe -> usearch_search) to:
 *   (a) build/search reference HNSW indexes for parity pinning, and
 *   (b) EXPORT the graph (entry slot, per-node levels, per-level adjacency)
 *       so the GPU product path can ingest the exact reference graph —
 *       the SURVEY.md §7.5 ingestion plan ("load the model with the usearch
 *       API and export vectors + per-node neighbor lists", not a parse of
 *       the serialized binary).
 *
 * Compiled against the usearch headers extracted from the vendored tarball
 * into oracle/_ref/src. `private` is remapped to reach index_dense_gt::typed_
 * and the adjacency tapes — acceptable here because this is checker-side
 * tooling, never product code.
 */

#include <algorithm>
#include <atomic>
#include <bitset>
#include <cassert>
#include <climits>
#include <cmath>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <iterator>
#include <map>
#include <memory>
#include <mutex>
#include <new>
#include <numeric>
#include <random>
#include <shared_mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <tuple>
#include <unordered_map>
#include <unordered_set>
#include <utility>
#include <vector>

#define private public
#include <usearch/index_dense.hpp>
#undef private

using namespace unum::usearch;

using dense_t = index_dense_t;  // index_dense_gt<u64 keys, u32 slots>

extern "C" {

struct ref_hnsw_t {
    dense_t index;
    std::string err;
};

static void set_err(char** err, const std::string& msg) {
    if (err) *err = strdup(msg.c_str());
}

/* metric: 0=l2sq, 1=ip (usearch: 1 - a.b), 2=cos (usearch: 1 - cossim) —
 * OpTypeToUsearchMetric (pkg/vectorindex/metric/types.go:152-163) */
void* ref_hnsw_create(size_t dim, int metric, size_t connectivity,
                      size_t expansion_add, size_t expansion_search,
                      size_t capacity, size_t threads, char** err) {
    metric_kind_t mk = metric == 0 ? metric_kind_t::l2sq_k
                      : metric == 1 ? metric_kind_t::ip_k
                                    : metric_kind_t::cos_k;
    metric_punned_t m(dim, mk, scalar_kind_t::f32_k);
    index_dense_config_t cfg(connectivity, expansion_add, expansion_search);
    auto state = dense_t::make(m, cfg);
    if (!state) {
        set_err(err, state.error.release());
        return nullptr;
    }
    auto h = new ref_hnsw_t{std::move(state.index), {}};
    index_limits_t limits(capacity, threads ? threads : 1);
    if (!h->index.try_reserve(limits)) {
        set_err(err, "reserve failed");
        delete h;
        return nullptr;
    }
    return h;
}

int ref_hnsw_add_batch(void* hp, const uint64_t* keys, const float* vecs,
                       size_t n, size_t dim, size_t threads, char** err) {
    auto h = (ref_hnsw_t*)hp;
    if (threads <= 1) {
        for (size_t i = 0; i < n; ++i) {
            auto r = h->index.add(keys[i], vecs + i * dim, 0);
            if (!r) { set_err(err, r.error.release()); return -1; }
        }
        return 0;
    }
    std::vector<std::thread> ts;
    std::vector<std::string> errs(threads);
    for (size_t t = 0; t < threads; ++t)
        ts.emplace_back([&, t] {
            for (size_t i = t; i < n; i += threads) {
                auto r = h->index.add(keys[i], vecs + i * dim, t);
                if (!r) { errs[t] = r.error.release(); return; }
            }
        });
    for (auto& t : ts) t.join();
    for (auto& e : errs)
        if (!e.empty()) { set_err(err, e); return -1; }
    return 0;
}

size_t ref_hnsw_search(void* hp, const float* query, size_t ef, size_t k,
                       uint64_t* out_keys, float* out_dists, char** err) {
    auto h = (ref_hnsw_t*)hp;
    h->index.change_expansion_search(ef);
    auto r = h->index.search(query, k);
    if (!r) { set_err(err, r.error.release()); return 0; }
    return r.dump_to(out_keys, out_dists);
}

/* filtered search: keep a candidate only when its KEY's bit is set —
 * index_dense passes the KEY to the predicate (index_dense.hpp:2245-2247),
 * exactly the shape of usearchex_filtered_search_with_membership's
 * per-key docfilter callback (cgo/usearchex.c:27-70). */
size_t ref_hnsw_search_filtered(void* hp, const float* query, size_t ef,
                                size_t k, const uint32_t* key_bits,
                                uint64_t nbits, uint64_t* out_keys,
                                float* out_dists, char** err) {
    auto h = (ref_hnsw_t*)hp;
    h->index.change_expansion_search(ef);
    auto pred = [key_bits, nbits](uint64_t key) {
        if (key >= nbits) return false;
        return ((key_bits[key >> 5] >> (key & 31)) & 1u) != 0;
    };
    auto r = h->index.filtered_search(query, k, pred);
    if (!r) { set_err(err, r.error.release()); return 0; }
    return r.dump_to(out_keys, out_dists);
}

size_t ref_hnsw_size(void* hp) { return ((ref_hnsw_t*)hp)->index.size(); }

int ref_hnsw_max_level(void* hp) {
    return (int)((ref_hnsw_t*)hp)->index.typed_->max_level();
}

uint64_t ref_hnsw_entry_slot(void* hp) {
    return (uint64_t)((ref_hnsw_t*)hp)->index.typed_->entry_slot_;
}

uint64_t ref_hnsw_key_of_slot(void* hp, uint32_t slot) {
    auto& typed = *((ref_hnsw_t*)hp)->index.typed_;
    return (uint64_t)typed.node_at_(slot).ckey();
}

int ref_hnsw_node_level(void* hp, uint32_t slot) {
    auto& typed = *((ref_hnsw_t*)hp)->index.typed_;
    return (int)typed.node_at_(slot).level();
}

/* neighbors of `slot` at `level` -> out (slot ids); returns count */
size_t ref_hnsw_neighbors(void* hp, uint32_t slot, int level, uint32_t* out,
                          size_t cap) {
    auto& typed = *((ref_hnsw_t*)hp)->index.typed_;
    auto node = typed.node_at_(slot);
    if (level > (int)node.level()) return 0;
    auto nbrs = level == 0
        ? typed.neighbors_base_(node)
        : typed.neighbors_non_base_(
              node, (std::remove_reference_t<decltype(typed)>::level_t)level);
    size_t n = nbrs.size();
    for (size_t i = 0; i < n && i < cap; ++i) out[i] = (uint32_t)nbrs[i];
    return n;
}

/* stored (possibly cast) vector of `slot` as f32 */
int ref_hnsw_get_vector(void* hp, uint32_t slot, float* out, size_t dim) {
    auto h = (ref_hnsw_t*)hp;
    const byte_t* v = h->index.vectors_lookup_[slot];
    if (!v) return -1;
    memcpy(out, v, dim * sizeof(float));
    return 0;
}

/* Bulk export: fills levels[n], keys[n], vecs[n*dim] and the level-0 CSR in
 * one call; upper levels via ref_hnsw_export_level. Returns l0 nnz. */
int64_t ref_hnsw_export_bulk(void* hp, size_t dim, int32_t* levels,
                             uint64_t* keys, float* vecs, int64_t* l0_offs,
                             uint32_t* l0_nbrs, int64_t l0_cap) {
    auto h = (ref_hnsw_t*)hp;
    auto& typed = *h->index.typed_;
    const size_t n = h->index.size();
    int64_t off = 0;
    for (size_t s = 0; s < n; ++s) {
        auto node = typed.node_at_((uint32_t)s);
        levels[s] = (int32_t)node.level();
        keys[s] = (uint64_t)node.ckey();
        const byte_t* v = h->index.vectors_lookup_[s];
        memcpy(vecs + s * dim, v, dim * sizeof(float));
        l0_offs[s] = off;
        auto nbrs = typed.neighbors_base_(node);
        const size_t cnt = nbrs.size();
        if (off + (int64_t)cnt > l0_cap) return -1;
        for (size_t i = 0; i < cnt; ++i) l0_nbrs[off++] = (uint32_t)nbrs[i];
    }
    l0_offs[n] = off;
    return off;
}

/* CSR for one upper level (1-based). Returns nnz or -1 on overflow. */
int64_t ref_hnsw_export_level(void* hp, int level, int64_t* offs,
                              uint32_t* nbrs, int64_t cap) {
    auto h = (ref_hnsw_t*)hp;
    auto& typed = *h->index.typed_;
    const size_t n = h->index.size();
    int64_t off = 0;
    using lvl_t = std::remove_reference_t<decltype(typed)>::level_t;
    for (size_t s = 0; s < n; ++s) {
        auto node = typed.node_at_((uint32_t)s);
        offs[s] = off;
        if ((int)node.level() >= level) {
            auto nb = typed.neighbors_non_base_(node, (lvl_t)level);
            const size_t cnt = nb.size();
            if (off + (int64_t)cnt > cap) return -1;
            for (size_t i = 0; i < cnt; ++i) nbrs[off++] = (uint32_t)nb[i];
        }
    }
    offs[n] = off;
    return off;
}

int ref_hnsw_save(void* hp, const char* path, char** err) {
    auto h = (ref_hnsw_t*)hp;
    auto r = h->index.save(path);
    if (!r) { set_err(err, r.error.release()); return -1; }
    return 0;
}

void* ref_hnsw_load(const char* path, char** err) {
    auto state = dense_t::make(path, /*view=*/false);
    if (!state) { set_err(err, state.error.release()); return nullptr; }
    return new ref_hnsw_t{std::move(state.index), {}};
}

void ref_hnsw_free(void* hp) { delete (ref_hnsw_t*)hp; }

}  // extern "C"
