/* CDNA4 HNSW search kernel — the GPU replacement for usearch's graph
 * traversal behind pkg/vectorindex/hnsw (model.go:881-901 SearchUnsafe ->
 * usearch_search; algorithm restated from the vendored usearch-2.25.3
 * index.hpp: search_for_one_ greedy descent :4466-4506 and
 * search_to_find_in_base_ beam :4688-4758 — see oracle/hnsw.py for the
 * CPU restatement both are parity-tested against).
 *
 * One workgroup (4 waves) per query. The ef-bounded candidate set lives in
 * LDS as a SORTED list of (dist_key32 | expanded_bit | slot31) u64 keys —
 * provably equivalent to usearch's next/top heap pair because every slot
 * pushed to `next` is also inserted into `top` under the same condition, so
 * "expand the best unexpanded list entry until none remain" visits exactly
 * the nodes usearch pops before its radius break. Distances are computed
 * wave-parallel (one row per wave, lanes split dims, shuffle reduce);
 * ~12.6 MB of random 3 KB row reads per query make this
 * latency-bound — concurrency comes from one workgroup per query.
 *
 * Metric values use usearch's conventions (l2sq; ip = 1 - a.b;
 * cos = 1 - cossim): the MO-side transform (DistanceTransformHnsw,
 * metric/types.go:234-243) is applied by the host layer above.
 */

#include <hip/hip_runtime.h>
#include <cfloat>
#include <climits>
#include <cstdint>

#include "moann_internal.h"

namespace moann {

enum HnswMetric : int { U_L2SQ = 0, U_IP = 1, U_COS = 2 };

__device__ __forceinline__ uint32_t hf2u(float f) {
    uint32_t u = __float_as_uint(f);
    return u ^ (((int32_t)u >> 31) | 0x80000000u);
}
__device__ __forceinline__ float hu2f(uint32_t u) {
    u ^= (((int32_t)(u ^ 0x80000000u)) >> 31) | 0x80000000u;
    return __uint_as_float(u);
}

#define EXPANDED_BIT 0x80000000ull
#define SLOT_MASK 0x7FFFFFFFull

/* one wave computes the metric distance of one row vs the LDS query */
template <int METRIC>
__device__ float wave_dist(const float* __restrict__ vecs, int64_t slot,
                           int dpad, const float* ldsq, float qnorm) {
    const int lane = threadIdx.x & 63;
    const float4* row = (const float4*)(vecs + slot * (int64_t)dpad);
    const int nquads = dpad >> 2;
    float dot = 0.f, nrm = 0.f;
    for (int j = lane; j < nquads; j += 64) {
        const float4 x = row[j];
        const float4 qv = ((const float4*)ldsq)[j];
        if (METRIC == U_L2SQ) {
            float e;
            e = x.x - qv.x; dot = fmaf(e, e, dot);
            e = x.y - qv.y; dot = fmaf(e, e, dot);
            e = x.z - qv.z; dot = fmaf(e, e, dot);
            e = x.w - qv.w; dot = fmaf(e, e, dot);
        } else {
            dot = fmaf(x.x, qv.x, dot);
            dot = fmaf(x.y, qv.y, dot);
            dot = fmaf(x.z, qv.z, dot);
            dot = fmaf(x.w, qv.w, dot);
            if (METRIC == U_COS)
                nrm += x.x * x.x + x.y * x.y + x.z * x.z + x.w * x.w;
        }
    }
#pragma unroll
    for (int w = 32; w; w >>= 1) {
        dot += __shfl_down(dot, w, 64);
        if (METRIC == U_COS) nrm += __shfl_down(nrm, w, 64);
    }
    if (METRIC == U_L2SQ) return dot;
    if (METRIC == U_IP) return 1.0f - dot;
    double denom = sqrt((double)nrm) * sqrt((double)qnorm);
    if (denom == 0.0) return 1.0f;
    double sim = (double)dot / denom;
    sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
    return (float)(1.0 - sim);
}

template <int METRIC>
__global__ __launch_bounds__(256) void hnsw_search_kernel(
    const float* __restrict__ vecs, int dpad, int64_t n,
    const int64_t* __restrict__ l0_offs, const uint32_t* __restrict__ l0_nbrs,
    const int32_t* __restrict__ u_idx, int64_t u_count,
    const int32_t* __restrict__ up_offs, const uint32_t* __restrict__ up_nbrs,
    int max_level, int64_t entry_slot,
    const float* __restrict__ queries, const float* __restrict__ qnorms,
    int ef, int k, int maxdeg_p2,
    uint32_t* __restrict__ visited, int64_t vwords,
    const int64_t* __restrict__ key_of_slot,
    int64_t* __restrict__ out_ids, float* __restrict__ out_dists,
    unsigned long long* __restrict__ eval_count) {
    extern __shared__ unsigned char smem[];
    const int q = blockIdx.x;
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;

    float* ldsq = (float*)smem;                               /* [dpad]    */
    uint64_t* list = (uint64_t*)(smem + ((dpad * 4 + 7) & ~7));/* [ef]     */
    uint64_t* merged = list + ef;                             /* [ef]      */
    uint64_t* newk = merged + ef;                             /* [maxdegP2]*/
    uint32_t* stage = (uint32_t*)(newk + maxdeg_p2);          /* [maxdegP2]*/
    float* wdist = (float*)(stage + maxdeg_p2);               /* [8]       */
    /* scalars */
    int* sc = (int*)(wdist + 8); /* 0:list_sz 1:nbr_idx 2:changed 3:n_new
                                    4:best_idx 5:evals 6:best_d(float) */

    const float* qsrc = queries + (int64_t)q * dpad;
    for (int e = tid; e < dpad; e += blockDim.x) ldsq[e] = qsrc[e];
    const float qn = (METRIC == U_COS) ? qnorms[q] : 0.f;
    uint32_t* vis = visited + (int64_t)q * vwords;
    if (tid == 0) sc[5] = 0;
    __syncthreads();

    /* ---- greedy descent over levels max..1 (search_for_one_) ----
     * one do-while pass over the current node's neighbor list takes the
     * overall min (identical to usearch's running-update scan: the end-of-
     * scan closest is the min over old closest and every candidate). */
    int64_t cur = entry_slot;
    float* best_d = (float*)&sc[6];
    {
        const float d0 = wave_dist<METRIC>(vecs, cur, dpad, ldsq, qn);
        if (tid == 0) *best_d = d0; /* identical in every wave */
    }
    __syncthreads();
    for (int level = max_level; level >= 1; --level) {
        for (;;) {
            const int32_t ui = u_idx[cur];
            const int32_t o0 = up_offs[(int64_t)(level - 1) * (u_count + 1) + ui];
            const int32_t o1 = up_offs[(int64_t)(level - 1) * (u_count + 1) + ui + 1];
            const int m = o1 - o0;
            if (tid == 0) sc[2] = 0;
            __syncthreads();
            for (int r = 0; r < m; r += 4) {
                const int i = r + wave;
                float d = 0.f;
                if (i < m)
                    d = wave_dist<METRIC>(vecs, (int64_t)up_nbrs[o0 + i],
                                          dpad, ldsq, qn);
                if (i < m && lane == 0) wdist[wave] = d;
                __syncthreads();
                if (tid == 0) {
                    const int lim = min(4, m - r);
                    for (int j = 0; j < lim; ++j) {
                        if (wdist[j] < *best_d) {
                            *best_d = wdist[j];
                            sc[1] = o0 + r + j; /* index into up_nbrs */
                            sc[2] = 1;
                        }
                    }
                }
                __syncthreads();
            }
            if (!sc[2]) break;
            cur = (int64_t)up_nbrs[sc[1]];
            __syncthreads();
        }
    }
    const float cur_d = *best_d;

    /* ---- base-level beam (search_to_find_in_base_) ---- */
    /* init: visited(entry), list = {entry} */
    if (tid == 0) {
        const int64_t w = cur >> 5;
        vis[w] |= (1u << (cur & 31));
        list[0] = ((uint64_t)hf2u(cur_d) << 32) | (uint64_t)cur;
        sc[0] = 1;
    }
    __syncthreads();

    const int max_expansions = 16 * ef + 64; /* bounded spin (guide rule) */
    for (int it = 0; it < max_expansions; ++it) {
        /* find best unexpanded entry (list ascending => first unexpanded) */
        if (tid == 0) sc[4] = INT_MAX;
        __syncthreads();
        const int sz = sc[0];
        for (int i = tid; i < sz; i += blockDim.x)
            if (!(list[i] & EXPANDED_BIT)) atomicMin(&sc[4], i);
        __syncthreads();
        const int bi = sc[4];
        if (bi == INT_MAX) break;
        const int64_t c_slot = (int64_t)(list[bi] & SLOT_MASK);
        if (tid == 0) list[bi] |= EXPANDED_BIT;
        __syncthreads();

        /* stage unvisited neighbors */
        const int64_t a0 = l0_offs[c_slot], a1 = l0_offs[c_slot + 1];
        const int m = (int)(a1 - a0);
        if (tid == 0) sc[3] = 0;
        __syncthreads();
        for (int i = tid; i < m; i += blockDim.x) {
            const uint32_t nb = l0_nbrs[a0 + i];
            const uint32_t old = atomicOr(&vis[nb >> 5], 1u << (nb & 31));
            if (!(old & (1u << (nb & 31)))) {
                const int pos = atomicAdd(&sc[3], 1);
                stage[pos] = nb;
            }
        }
        __syncthreads();
        const int mp = sc[3];
        if (tid == 0) atomicAdd(&sc[5], mp);

        /* distances: 4 waves round-robin over the staged slots */
        for (int r = 0; r < mp; r += 4) {
            const int i = r + wave;
            float d = 0.f;
            if (i < mp)
                d = wave_dist<METRIC>(vecs, (int64_t)stage[i], dpad, ldsq, qn);
            if (i < mp && lane == 0)
                newk[i] = ((uint64_t)hf2u(d) << 32) | (uint64_t)stage[i];
            __syncthreads();
        }
        for (int i = mp + tid; i < maxdeg_p2; i += blockDim.x)
            newk[i] = ~0ull;
        __syncthreads();

        /* bitonic sort of the <=maxdeg_p2 new keys */
        for (int ks = 2; ks <= maxdeg_p2; ks <<= 1) {
            for (int jj = ks >> 1; jj > 0; jj >>= 1) {
                for (int i = tid; i < maxdeg_p2; i += blockDim.x) {
                    const int l = i ^ jj;
                    if (l > i) {
                        const uint64_t a = newk[i], b = newk[l];
                        const bool up = (i & ks) == 0;
                        if ((a > b) == up) { newk[i] = b; newk[l] = a; }
                    }
                }
                __syncthreads();
            }
        }

        /* merge-path: C = first min(sz+mp, ef) of merge(list[0..sz), newk[0..mp)) */
        const int outn = min(sz + mp, ef);
        for (int i = tid; i < outn; i += blockDim.x) {
            int lo = max(0, i - mp), hi = min(i, sz);
            while (lo < hi) {
                const int mid = (lo + hi) >> 1;
                if ((list[mid] & ~EXPANDED_BIT) <=
                    (newk[i - mid - 1] & ~EXPANDED_BIT))
                    lo = mid + 1;
                else
                    hi = mid;
            }
            const int a = lo, b = i - lo;
            const uint64_t va = a < sz ? list[a] : ~0ull;
            const uint64_t vb = b < mp ? newk[b] : ~0ull;
            merged[i] = (va & ~EXPANDED_BIT) <= (vb & ~EXPANDED_BIT) ? va : vb;
        }
        __syncthreads();
        for (int i = tid; i < outn; i += blockDim.x) list[i] = merged[i];
        if (tid == 0) sc[0] = outn;
        __syncthreads();
    }

    /* output first k (ascending), padded */
    const int sz = sc[0];
    for (int i = tid; i < k; i += blockDim.x) {
        if (i < sz) {
            const uint64_t kv = list[i];
            out_ids[(int64_t)q * k + i] = key_of_slot[kv & SLOT_MASK];
            out_dists[(int64_t)q * k + i] = hu2f((uint32_t)(kv >> 32));
        } else {
            out_ids[(int64_t)q * k + i] = -1;
            out_dists[(int64_t)q * k + i] = FLT_MAX;
        }
    }
    if (tid == 0 && eval_count) atomicAdd(eval_count, (unsigned long long)sc[5]);
}

void launch_hnsw_search(int metric, const float* vecs, int dpad, int64_t n,
                        const int64_t* l0_offs, const uint32_t* l0_nbrs,
                        const int32_t* u_idx, int64_t u_count,
                        const int32_t* up_offs, const uint32_t* up_nbrs,
                        int max_level, int64_t entry_slot,
                        const float* queries, const float* qnorms, int nq,
                        int ef, int k, int maxdeg_p2, uint32_t* visited,
                        int64_t vwords, const int64_t* key_of_slot,
                        int64_t* out_ids, float* out_dists,
                        unsigned long long* eval_count, size_t* shmem_out,
                        hipStream_t stream) {
    const size_t shmem = ((dpad * 4 + 7) & ~7) + (size_t)ef * 8 * 2 +
                         (size_t)maxdeg_p2 * 8 + (size_t)maxdeg_p2 * 4 +
                         8 * 4 + 8 * 4 + 16;
    if (shmem_out) *shmem_out = shmem;
    if (!nq) return;
#define HARGS vecs, dpad, n, l0_offs, l0_nbrs, u_idx, u_count, up_offs,     \
    up_nbrs, max_level, entry_slot, queries, qnorms, ef, k, maxdeg_p2,      \
    visited, vwords, key_of_slot, out_ids, out_dists, eval_count
    switch (metric) {
    case U_L2SQ:
        hipLaunchKernelGGL((hnsw_search_kernel<U_L2SQ>), dim3(nq), dim3(256),
                           shmem, stream, HARGS);
        break;
    case U_IP:
        hipLaunchKernelGGL((hnsw_search_kernel<U_IP>), dim3(nq), dim3(256),
                           shmem, stream, HARGS);
        break;
    default:
        hipLaunchKernelGGL((hnsw_search_kernel<U_COS>), dim3(nq), dim3(256),
                           shmem, stream, HARGS);
        break;
    }
#undef HARGS
}

}  // namespace moann
