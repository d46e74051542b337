/* libmoann_hip — HNSW host side (see include/moann.h HNSW section).
 * Ingests a usearch-exported graph into HBM and drives the CDNA4 beam-search
 * kernel (hnsw_kernels.hip). Same conventions as the IVF side: errmsg
 * out-params (cgo/cuvs/helper.h:59), grow-only workspaces, fail-loud. */

#include <hip/hip_runtime.h>

#include <cfloat>
#include <algorithm>
#include <cstring>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

#include "../../include/moann.h"
#include "moann_internal.h"

namespace moann {
void launch_hnsw_search(int metric, const float* vecs, int dpad, int64_t n,
                        const int64_t* l0_offs, const uint32_t* l0_nbrs,
                        const int32_t* u_idx, int64_t u_count,
                        const int32_t* up_offs, const uint32_t* up_nbrs,
                        int max_level, int64_t entry_slot,
                        const float* queries, const float* qnorms, int nq,
                        int ef, int k, int maxdeg_p2,
                        const uint32_t* slot_filter, uint32_t* visited,
                        int64_t vwords, const int64_t* key_of_slot,
                        int64_t* out_ids, float* out_dists,
                        unsigned long long* eval_count, size_t* shmem_out,
                        hipStream_t stream);
}

namespace {

using namespace moann;

#define HIP_CHECK(expr)                                                     \
    do {                                                                    \
        hipError_t _e = (expr);                                             \
        if (_e != hipSuccess)                                               \
            throw std::runtime_error(std::string("HIP error: ") +           \
                                     hipGetErrorString(_e) + " at " #expr); \
    } while (0)

void hn_set_errmsg(void* errmsg, const char* context, const char* message) {
    if (!errmsg) return;
    std::string full = std::string(context) + ": " + message;
    *(char**)errmsg = strdup(full.c_str());
}

struct HnswDevBuf {
    void* ptr = nullptr;
    size_t bytes = 0;
    void ensure(size_t need) {
        if (need <= bytes) return;
        if (ptr) (void)hipFree(ptr);
        ptr = nullptr;
        bytes = 0;
        HIP_CHECK(hipMalloc(&ptr, need));
        bytes = need;
    }
    ~HnswDevBuf() {
        if (ptr) (void)hipFree(ptr);
    }
    template <typename T>
    T* as() { return (T*)ptr; }
};

struct HnswIndex {
    int device = 0;
    uint32_t dim = 0, dpad = 0;
    int metric = 0; /* usearch kind: 0 l2sq, 1 ip, 2 cos */
    uint64_t n = 0;
    int64_t entry_slot = 0;
    int32_t max_level = 0;
    int maxdeg = 0;
    bool built = false;
    bool have_vecs = false, have_l0 = false, have_upper = false;

    hipStream_t stream = nullptr;
    float* d_vecs = nullptr;        /* [n][dpad] */
    int64_t* d_l0_offs = nullptr;
    uint32_t* d_l0_nbrs = nullptr;
    int32_t* d_u_idx = nullptr;
    int64_t u_count = 0;
    int32_t* d_up_offs = nullptr;
    uint32_t* d_up_nbrs = nullptr;
    int64_t* d_keys = nullptr;
    std::vector<int64_t> h_keys; /* slot -> key, for docfilter bitsets */
    uint32_t default_ef = 64;    /* moann_hnsw_set_ef (usearchex entry) */

    HnswDevBuf w_queries, w_qnorms, w_visited, w_out_ids, w_out_dists,
        w_evals, w_filter;

    std::mutex mu;
    /* perf: whole-kernel time under scan_*, evals -> bytes */
    moann_perf_t perf {};
    hipEvent_t ev[2] {};
    bool ev_made = false;

    ~HnswIndex() {
        (void)hipSetDevice(device);
        for (auto p : {(void*)d_vecs, (void*)d_l0_offs, (void*)d_l0_nbrs,
                       (void*)d_u_idx, (void*)d_up_offs, (void*)d_up_nbrs,
                       (void*)d_keys})
            if (p) (void)hipFree(p);
        if (ev_made)
            for (auto& e : ev) (void)hipEventDestroy(e);
        if (stream) (void)hipStreamDestroy(stream);
    }
};

HnswIndex* HX(moann_hnsw_c h) { return (HnswIndex*)h; }

void run_hnsw_search(HnswIndex* ix, const float* queries, bool on_device,
                     uint64_t nq, uint32_t ef, uint32_t k, int64_t* out_keys,
                     float* out_dists,
                     const uint32_t* slot_filter_host = nullptr) {
    std::lock_guard<std::mutex> lk(ix->mu);
    if (!ix->built) throw std::runtime_error("index not built");
    if (nq == 0) return;
    if (ef == 0) ef = 1;
    if (ef < k) ef = k; /* expansion = max(config.expansion, wanted),
                           index.hpp:3475 */
    if (ef > 1024) throw std::runtime_error("ef > 1024 unsupported");
    HIP_CHECK(hipSetDevice(ix->device));
    const hipStream_t s = ix->stream;
    const int dpad = (int)ix->dpad;
    if (!ix->ev_made) {
        for (auto& e : ix->ev) HIP_CHECK(hipEventCreate(&e));
        ix->ev_made = true;
    }

    ix->w_queries.ensure((size_t)nq * dpad * 4);
    if ((int)ix->dim != dpad)
        HIP_CHECK(hipMemsetAsync(ix->w_queries.ptr, 0, (size_t)nq * dpad * 4, s));
    HIP_CHECK(hipMemcpy2DAsync(ix->w_queries.ptr, (size_t)dpad * 4, queries,
                               (size_t)ix->dim * 4, (size_t)ix->dim * 4, nq,
                               on_device ? hipMemcpyDeviceToDevice
                                         : hipMemcpyHostToDevice, s));
    const float* d_q = ix->w_queries.as<float>();
    const float* d_qn = nullptr;
    if (ix->metric == 2) {
        ix->w_qnorms.ensure(nq * 4);
        launch_qnorms(d_q, (int)nq, dpad, ix->w_qnorms.as<float>(), s);
        d_qn = ix->w_qnorms.as<float>();
    }

    const int64_t vwords = (int64_t)((ix->n + 31) / 32);
    ix->w_visited.ensure((size_t)nq * vwords * 4);
    HIP_CHECK(hipMemsetAsync(ix->w_visited.ptr, 0, (size_t)nq * vwords * 4, s));
    ix->w_out_ids.ensure((size_t)nq * k * 8);
    ix->w_out_dists.ensure((size_t)nq * k * 4);
    ix->w_evals.ensure(8);
    HIP_CHECK(hipMemsetAsync(ix->w_evals.ptr, 0, 8, s));
    const uint32_t* d_filter = nullptr;
    if (slot_filter_host) {
        if (ix->n >= (1ull << 30))
            throw std::runtime_error("filtered search needs n < 2^30");
        const size_t fw = (size_t)((ix->n + 31) / 32);
        ix->w_filter.ensure(fw * 4);
        HIP_CHECK(hipMemcpyAsync(ix->w_filter.ptr, slot_filter_host, fw * 4,
                                 hipMemcpyHostToDevice, s));
        d_filter = ix->w_filter.as<uint32_t>();
    }

    int maxdeg_p2 = 64;
    while (maxdeg_p2 < ix->maxdeg) maxdeg_p2 <<= 1;

    size_t shmem = 0;
    HIP_CHECK(hipEventRecord(ix->ev[0], s));
    launch_hnsw_search(ix->metric, ix->d_vecs, dpad, (int64_t)ix->n,
                       ix->d_l0_offs, ix->d_l0_nbrs, ix->d_u_idx, ix->u_count,
                       ix->d_up_offs, ix->d_up_nbrs, ix->max_level,
                       ix->entry_slot, d_q, d_qn, (int)nq, (int)ef, (int)k,
                       maxdeg_p2, d_filter, ix->w_visited.as<uint32_t>(), vwords,
                       ix->d_keys, ix->w_out_ids.as<int64_t>(),
                       ix->w_out_dists.as<float>(),
                       ix->w_evals.as<unsigned long long>(), &shmem, s);
    {
        hipError_t e = hipGetLastError();
        if (e != hipSuccess)
            throw std::runtime_error(
                std::string("hnsw kernel launch [shmem ") +
                std::to_string(shmem) + "]: " + hipGetErrorString(e));
    }
    HIP_CHECK(hipEventRecord(ix->ev[1], s));
    unsigned long long evals = 0;
    HIP_CHECK(hipMemcpyAsync(out_keys, ix->w_out_ids.ptr, (size_t)nq * k * 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipMemcpyAsync(out_dists, ix->w_out_dists.ptr,
                             (size_t)nq * k * 4, hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipMemcpyAsync(&evals, ix->w_evals.ptr, 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(hipGetLastError());

    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, ix->ev[0], ix->ev[1]));
    ix->perf.scan_launches++;
    ix->perf.scan_ms += ms;
    ix->perf.scan_rows += evals;
    ix->perf.scan_bytes += evals * (uint64_t)ix->dim * 4;
}

}  // namespace

extern "C" {

moann_hnsw_c moann_hnsw_new(uint32_t dimension, int usearch_metric,
                            uint64_t n_nodes, int64_t entry_slot,
                            int32_t max_level, const int* devices,
                            int device_count, void* errmsg) {
    try {
        if (usearch_metric < 0 || usearch_metric > 2)
            throw std::runtime_error("usearch_metric must be 0/1/2");
        auto ix = std::make_unique<HnswIndex>();
        ix->device = (devices && device_count > 0) ? devices[0] : 0;
        HIP_CHECK(hipSetDevice(ix->device));
        HIP_CHECK(hipStreamCreate(&ix->stream));
        ix->dim = dimension;
        ix->dpad = (dimension + 3) & ~3u;
        ix->metric = usearch_metric;
        ix->n = n_nodes;
        ix->entry_slot = entry_slot;
        ix->max_level = max_level;
        return (moann_hnsw_c)ix.release();
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_new", e.what());
        return nullptr;
    }
}

void moann_hnsw_set_vectors(moann_hnsw_c h, const float* vecs, void* errmsg) {
    try {
        auto ix = HX(h);
        HIP_CHECK(hipSetDevice(ix->device));
        HIP_CHECK(hipMalloc(&ix->d_vecs, (size_t)ix->n * ix->dpad * 4));
        if (ix->dim != ix->dpad)
            HIP_CHECK(hipMemset(ix->d_vecs, 0, (size_t)ix->n * ix->dpad * 4));
        HIP_CHECK(hipMemcpy2D(ix->d_vecs, (size_t)ix->dpad * 4, vecs,
                              (size_t)ix->dim * 4, (size_t)ix->dim * 4, ix->n,
                              hipMemcpyHostToDevice));
        ix->have_vecs = true;
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_set_vectors", e.what());
    }
}

void moann_hnsw_set_keys(moann_hnsw_c h, const int64_t* keys, void* errmsg) {
    try {
        auto ix = HX(h);
        HIP_CHECK(hipSetDevice(ix->device));
        HIP_CHECK(hipMalloc(&ix->d_keys, (size_t)ix->n * 8));
        HIP_CHECK(hipMemcpy(ix->d_keys, keys, (size_t)ix->n * 8,
                            hipMemcpyHostToDevice));
        ix->h_keys.assign(keys, keys + ix->n);
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_set_keys", e.what());
    }
}

void moann_hnsw_set_level0(moann_hnsw_c h, const int64_t* offs,
                           const uint32_t* nbrs, int64_t nnz, void* errmsg) {
    try {
        auto ix = HX(h);
        HIP_CHECK(hipSetDevice(ix->device));
        HIP_CHECK(hipMalloc(&ix->d_l0_offs, (size_t)(ix->n + 1) * 8));
        HIP_CHECK(hipMemcpy(ix->d_l0_offs, offs, (size_t)(ix->n + 1) * 8,
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMalloc(&ix->d_l0_nbrs,
                            (size_t)(nnz > 0 ? nnz : 1) * 4));
        if (nnz)
            HIP_CHECK(hipMemcpy(ix->d_l0_nbrs, nbrs, (size_t)nnz * 4,
                                hipMemcpyHostToDevice));
        int maxdeg = 0;
        for (uint64_t i = 0; i < ix->n; ++i)
            maxdeg = std::max<int>(maxdeg, (int)(offs[i + 1] - offs[i]));
        ix->maxdeg = std::max(ix->maxdeg, maxdeg);
        ix->have_l0 = true;
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_set_level0", e.what());
    }
}

void moann_hnsw_set_upper(moann_hnsw_c h, const int32_t* u_idx,
                          int64_t u_count, const int32_t* up_offs,
                          const uint32_t* up_nbrs, int64_t up_nnz,
                          void* errmsg) {
    try {
        auto ix = HX(h);
        HIP_CHECK(hipSetDevice(ix->device));
        ix->u_count = u_count;
        HIP_CHECK(hipMalloc(&ix->d_u_idx, (size_t)ix->n * 4));
        HIP_CHECK(hipMemcpy(ix->d_u_idx, u_idx, (size_t)ix->n * 4,
                            hipMemcpyHostToDevice));
        const size_t offn = (size_t)ix->max_level * (u_count + 1);
        HIP_CHECK(hipMalloc(&ix->d_up_offs, std::max<size_t>(1, offn) * 4));
        if (offn)
            HIP_CHECK(hipMemcpy(ix->d_up_offs, up_offs, offn * 4,
                                hipMemcpyHostToDevice));
        HIP_CHECK(hipMalloc(&ix->d_up_nbrs,
                            (size_t)(up_nnz > 0 ? up_nnz : 1) * 4));
        if (up_nnz)
            HIP_CHECK(hipMemcpy(ix->d_up_nbrs, up_nbrs, (size_t)up_nnz * 4,
                                hipMemcpyHostToDevice));
        ix->have_upper = true;
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_set_upper", e.what());
    }
}

void moann_hnsw_build(moann_hnsw_c h, void* errmsg) {
    try {
        auto ix = HX(h);
        if (!ix->have_vecs || !ix->have_l0)
            throw std::runtime_error("vectors and level0 adjacency required");
        if (ix->max_level > 0 && !ix->have_upper)
            throw std::runtime_error("upper adjacency required (max_level>0)");
        if (!ix->d_keys) {
            /* identity keys */
            std::vector<int64_t> ids(ix->n);
            for (uint64_t i = 0; i < ix->n; ++i) ids[i] = (int64_t)i;
            moann_hnsw_set_keys(h, ids.data(), errmsg);
        }
        ix->built = true;
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_build", e.what());
    }
}

void moann_hnsw_search(moann_hnsw_c h, const float* queries, uint64_t nq,
                       uint32_t dim, uint32_t ef, uint32_t k,
                       int64_t* out_keys, float* out_dists, void* errmsg) {
    try {
        auto ix = HX(h);
        if (dim != ix->dim) throw std::runtime_error("query dim mismatch");
        run_hnsw_search(ix, queries, false, nq, ef, k, out_keys, out_dists);
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_search", e.what());
    }
}

void moann_hnsw_search_device(moann_hnsw_c h, const void* queries_dev,
                              uint64_t nq, uint32_t dim, uint32_t ef,
                              uint32_t k, int64_t* out_keys, float* out_dists,
                              void* errmsg) {
    try {
        auto ix = HX(h);
        if (dim != ix->dim) throw std::runtime_error("query dim mismatch");
        run_hnsw_search(ix, (const float*)queries_dev, true, nq, ef, k,
                        out_keys, out_dists);
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_search_device", e.what());
    }
}

/* Membership-filtered HNSW search — the GPU counterpart of
 * usearchex_filtered_search_with_membership (cgo/usearchex.c:51-70): the
 * membership test gates the reportable set exactly like usearch's predicate
 * (index.hpp:4712-4752). slot_bitset: one bit per SLOT; the host evaluates
 * its docfilter (bloom/croaring/cbitmap/sorted64 handle) over key_of_slot
 * once per filter, the cuVS eval_filter_bitmap_cpu shape. */
void moann_hnsw_search_filtered(moann_hnsw_c h, const float* queries,
                                uint64_t nq, uint32_t dim, uint32_t ef,
                                uint32_t k, const uint32_t* slot_bitset,
                                int64_t* out_keys, float* out_dists,
                                void* errmsg) {
    try {
        auto ix = HX(h);
        if (dim != ix->dim) throw std::runtime_error("query dim mismatch");
        if (!slot_bitset) throw std::runtime_error("null filter bitset");
        run_hnsw_search(ix, queries, false, nq, ef, k, out_keys, out_dists,
                        slot_bitset);
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_search_filtered", e.what());
    }
}

void moann_hnsw_set_ef(moann_hnsw_c h, uint32_t ef, void* errmsg) {
    try {
        if (ef == 0) throw std::runtime_error("ef must be positive");
        HX(h)->default_ef = ef;
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_set_ef", e.what());
    }
}

/* cgo/usearchex.h:38-42 / cgo/usearchex.c:51-70 — the filtered-search
 * bridge the Go side binds, on a moann_hnsw_c. The membership handle is a
 * moann_docfilter_c (moann_docfilter_parse of the serialized payload); it
 * is evaluated over the slot->key map into a slot bitset (the GPU-side
 * eval_filter_bitmap_cpu shape, cgo/cuvs/filter.hpp:855-960) and gates the
 * REPORTABLE set exactly like usearch's predicate. Returns the number of
 * results written; on failure writes a message into *error (mo_error_t
 * convention: a NUL-terminated string, usearchex.h:23; storage is
 * thread-local and stays valid until the next failing call on the
 * thread). */
size_t usearchex_filtered_search_with_membership(
    moann_hnsw_c index, const void* query_vector, int query_kind,
    size_t count, void* filter, int filter_kind, int64_t* keys,
    float* distances, const char** error) {
    static thread_local std::string err_store;
    try {
        auto ix = HX(index);
        if (query_kind != 1 /* usearch_scalar_f32_k, c/usearch.h:57 */)
            throw std::runtime_error(
                "query kind must be usearch_scalar_f32_k (MO searches f32, "
                "usearchex/search.go:68)");
        if (count == 0) return 0;
        const uint32_t ef =
            std::max<uint32_t>(ix->default_ef, (uint32_t)count);
        std::vector<uint32_t> bits;
        if (filter) {
            bits.assign((size_t)((ix->n + 31) / 32), 0);
            if (ix->h_keys.empty())
                throw std::runtime_error("keys not set on the index");
            char* inner = nullptr;
            if (moann_docfilter_test((moann_docfilter_c)filter,
                                     ix->h_keys.data(), ix->n, bits.data(),
                                     &inner) != 0) {
                std::string m = inner ? inner : "filter evaluation failed";
                if (inner) free(inner);
                (void)filter_kind;
                throw std::runtime_error(m);
            }
        }
        std::vector<int64_t> okeys(count);
        std::vector<float> odists(count);
        run_hnsw_search(ix, (const float*)query_vector, false, 1, ef,
                        (uint32_t)count, okeys.data(), odists.data(),
                        filter ? bits.data() : nullptr);
        size_t found = 0;
        for (size_t i = 0; i < count; ++i) {
            if (okeys[i] == -1) break;
            keys[found] = okeys[i];
            distances[found] = odists[i];
            ++found;
        }
        return found;
    } catch (const std::exception& e) {
        if (error) {
            err_store = e.what();
            *error = err_store.c_str();
        }
        return 0;
    }
}

void moann_hnsw_perf(moann_hnsw_c h, moann_perf_t* out) { *out = HX(h)->perf; }
void moann_hnsw_perf_reset(moann_hnsw_c h) { HX(h)->perf = moann_perf_t {}; }

void moann_hnsw_destroy(moann_hnsw_c h, void* errmsg) {
    try {
        delete HX(h);
    } catch (const std::exception& e) {
        hn_set_errmsg(errmsg, "moann_hnsw_destroy", e.what());
    }
}

}  /* extern "C" */
