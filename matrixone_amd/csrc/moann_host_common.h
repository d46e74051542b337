/* Shared host-side helpers for the moann C-ABI libraries (IVF-Flat, HNSW,
 * IVF-PQ). Internal — not part of the public ABI. */

#ifndef MOANN_HOST_COMMON_H
#define MOANN_HOST_COMMON_H

#include <hip/hip_runtime.h>

#include <cstring>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace moann {

#define MOANN_HIP_CHECK(expr)                                               \
    do {                                                                    \
        hipError_t _e = (expr);                                             \
        if (_e != hipSuccess)                                               \
            throw std::runtime_error(std::string("HIP error: ") +           \
                                     hipGetErrorString(_e) + " at " #expr); \
    } while (0)

#define MOANN_KCHECK(tag)                                                    \
    do {                                                                     \
        hipError_t _e = hipGetLastError();                                   \
        if (_e != hipSuccess)                                                \
            throw std::runtime_error(std::string("HIP launch error [") +     \
                                     tag + "]: " + hipGetErrorString(_e));   \
    } while (0)

/* errmsg convention: cgo/cuvs/helper.h:59 — errmsg is a char** as void*. */
inline void moann_set_errmsg(void* errmsg, const char* context,
                             const char* message) {
    if (!errmsg) return;
    std::string full = std::string(context) + ": " + message;
    *(char**)errmsg = strdup(full.c_str());
}

/* The result object behind gpu_ivf_flat_result_c — ALL index families
 * return this type so the gpu_ivf_flat_get_neighbors/get_distances/
 * free_result readout (ivf_flat_c.h:139-141) works for every search. */
struct MoannResult {
    std::vector<int64_t> neighbors;
    std::vector<float> distances;
    /* async pipeline (moann_ivf_flat_search_submit): while !collected the
     * payload lives in the owning SearchCtx's pinned bounce buffer and
     * done_ev marks the D2H completion on that ctx's stream. */
    void* owner = nullptr;        /* SearchCtx* (IVF lib) or null          */
    hipEvent_t done_ev = nullptr; /* owned by the result; null => sync    */
    bool collected = true;
    /* extend support: the tail sub-index's (already collected) result,
     * top-k merged into this one at collect (gpu_ivf_flat_extend) */
    std::unique_ptr<MoannResult> tail_res;
    int merge_kmetric = 0;   /* KM_* of the owning index (IP merge order) */
    uint32_t merge_limit = 0; /* row stride (limit) for the tail merge */
    bool tail_merged = false;
};

/* grow-only device buffer (cuvs_worker.hpp:50-150 shape, minus RMM) */
struct MoannDevBuf {
    void* ptr = nullptr;
    size_t bytes = 0;
    void ensure(size_t need) {
        if (need <= bytes) return;
        if (ptr) (void)hipFree(ptr);
        ptr = nullptr;
        bytes = 0;
        MOANN_HIP_CHECK(hipMalloc(&ptr, need));
        bytes = need;
    }
    ~MoannDevBuf() {
        if (ptr) (void)hipFree(ptr);
    }
    template <typename T>
    T* as() { return (T*)ptr; }
};

}  // namespace moann

#endif
