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
/* filtered variant: bit30 of the slot word = "passes the membership
 * filter"; slots then fit 30 bits (host rejects n >= 2^30) */
#define PASS_BIT 0x40000000ull
#define FSLOT_MASK 0x3FFFFFFFull

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

template <int METRIC, bool FILTERED>
__global__ __launch_bounds__(1024) void hnsw_search_kernel(
    const float* __restrict__ vecs, int dpad, int64_t n,
    const int64_t* __restrict__ l0_offs, const uint32_t* __restrict__ l0_nbrs,
    const int32_t* __restrict__ u_idx, int64_t u_count,
    const int32_t* __restrict__ up_offs, const uint32_t* __restrict__ up_nbrs,
    int max_level, int64_t entry_slot,
    const float* __restrict__ queries, const float* __restrict__ qnorms,
    int ef, int efc /* list capacity: ef, or 2*ef when FILTERED */,
    int k, int maxdeg_p2,
    const uint32_t* __restrict__ slot_filter, /* FILTERED only: bit/slot */
    uint32_t* __restrict__ visited, int64_t vwords,
    const int64_t* __restrict__ key_of_slot,
    int64_t* __restrict__ out_ids, float* __restrict__ out_dists,
    unsigned long long* __restrict__ eval_count) {
    extern __shared__ unsigned char smem[];
    const int q = blockIdx.x;
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int nwave = (int)blockDim.x >> 6; /* 16 at the 1024-thread launch:
        neighbor evaluations are the latency bottleneck (random 3 KB row
        reads), so width = parallel evals per expansion round */

    float* ldsq = (float*)smem;                               /* [dpad]    */
    uint64_t* list = (uint64_t*)(smem + ((dpad * 4 + 7) & ~7));/* [efc]    */
    uint64_t* merged = list + efc;                            /* [efc]     */
    uint64_t* newk = merged + efc;                            /* [maxdegP2]*/
    uint32_t* stage = (uint32_t*)(newk + maxdeg_p2);          /* [maxdegP2]*/
    float* wdist = (float*)(stage + maxdeg_p2);               /* [16]      */
    /* scalars */
    int* sc = (int*)(wdist + 16); /* 0:list_sz 1:nbr_idx 2:kept 3:n_new
                                    4:best_idx 5:evals 6:best_d/radius
                                    7:pass_cnt */
    int* cnt_scratch = sc + 8;                            /* [1024] */

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
            for (int r = 0; r < m; r += nwave) {
                const int i = r + wave;
                float d = 0.f;
                if (i < m)
                    d = wave_dist<METRIC>(vecs, (int64_t)up_nbrs[o0 + i],
                                          dpad, ldsq, qn);
                if (i < m && lane == 0) wdist[wave] = d;
                __syncthreads();
                if (tid == 0) {
                    const int lim = min(nwave, m - r);
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

    /* ---- base-level beam (search_to_find_in_base_) ----
     * FILTERED follows usearch's predicate semantics exactly
     * (index.hpp:4712-4752): the membership test gates entry into `top`
     * (= the reportable set and the radius) while traversal continues
     * through filtered nodes; the list holds BOTH kinds, flagged, with
     * capacity efc = 2*ef (entries past efc are dropped — they are farther
     * than the 2ef-th best candidate; usearch's unbounded `next` would keep
     * them, a bounded-memory deviation that matters only when more than ef
     * filtered candidates interleave the passing front). */
    const auto passes = [&](int64_t slot) -> bool {
        if (!FILTERED) return true;
        return (slot_filter[slot >> 5] >> (slot & 31)) & 1u;
    };
    if (tid == 0) {
        const int64_t w = cur >> 5;
        vis[w] |= (1u << (cur & 31));
        uint64_t kv = ((uint64_t)hf2u(cur_d) << 32) | (uint64_t)cur;
        if (FILTERED && passes(cur)) kv |= PASS_BIT;
        list[0] = kv;
        sc[0] = 1;
    }
    __syncthreads();
    const uint64_t slot_mask = FILTERED ? FSLOT_MASK : SLOT_MASK;

    const int max_expansions = 16 * efc + 64; /* bounded spin (guide rule) */
    for (int it = 0; it < max_expansions; ++it) {
        const int sz = sc[0];
        /* FILTERED: passing count + the ef-th passing dist (= radius).
         * contiguous per-thread blocks so thread 0 can localise the ef-th. */
        if (FILTERED) {
            const int per = (sz + blockDim.x - 1) / blockDim.x;
            int cnt = 0;
            for (int i = tid * per; i < min((tid + 1) * per, sz); ++i)
                if (list[i] & PASS_BIT) cnt++;
            cnt_scratch[tid] = cnt;
            __syncthreads();
            if (tid == 0) {
                int acc = 0, radius_bits = (int)0xFFFFFFFFu; /* +inf key */
                int total = 0;
                for (int t = 0; t < (int)blockDim.x; ++t)
                    total += cnt_scratch[t];
                for (int t = 0; t < (int)blockDim.x && acc + 1 <= ef; ++t) {
                    const int c = cnt_scratch[t];
                    if (acc + c >= ef) {
                        int need = ef - acc;
                        for (int i = t * per; i < min((t + 1) * per, sz);
                             ++i)
                            if (list[i] & PASS_BIT && --need == 0) {
                                radius_bits = (int)(list[i] >> 32);
                                break;
                            }
                        acc = ef;
                        break;
                    }
                    acc += c;
                }
                sc[6] = radius_bits; /* u32 key of the radius */
                sc[7] = total;       /* passing count */
            }
            __syncthreads();
        }
        const uint32_t radius_key = FILTERED ? (uint32_t)sc[6] : 0;
        const int pass_cnt = FILTERED ? sc[7] : 0;

        /* find best unexpanded entry (list ascending => first unexpanded) */
        if (tid == 0) sc[4] = INT_MAX;
        __syncthreads();
        for (int i = tid; i < sz; i += blockDim.x)
            if (!(list[i] & EXPANDED_BIT)) atomicMin(&sc[4], i);
        __syncthreads();
        const int bi = sc[4];
        if (bi == INT_MAX) break;
        if (FILTERED && pass_cnt >= ef &&
            (uint32_t)(list[bi] >> 32) > radius_key)
            break; /* usearch: candidate.dist > radius && top full */
        const int64_t c_slot = (int64_t)(list[bi] & slot_mask);
        if (tid == 0) list[bi] |= EXPANDED_BIT;
        __syncthreads();

        /* stage unvisited neighbors */
        const int64_t a0 = l0_offs[c_slot], a1 = l0_offs[c_slot + 1];
        const int m = (int)(a1 - a0);
        if (tid == 0) { sc[3] = 0; sc[2] = 0; }
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

        /* distances: all waves round-robin; FILTERED drops candidates
         * that fail usearch's push condition (top full && d >= radius) and
         * tags the rest with the membership bit */
        for (int r = 0; r < mp; r += nwave) {
            const int i = r + wave;
            float d = 0.f;
            if (i < mp)
                d = wave_dist<METRIC>(vecs, (int64_t)stage[i], dpad, ldsq, qn);
            if (i < mp && lane == 0) {
                const uint32_t dk = hf2u(d);
                if (FILTERED && pass_cnt >= ef && dk >= radius_key) {
                    /* not pushed (neither next nor top) */
                } else {
                    const int pos = atomicAdd(&sc[2], 1);
                    uint64_t kv = ((uint64_t)dk << 32) | (uint64_t)stage[i];
                    if (FILTERED && passes((int64_t)stage[i])) kv |= PASS_BIT;
                    newk[pos] = kv;
                }
            }
            __syncthreads();
        }
        const int kept = sc[2]; /* == mp when unfiltered */
        for (int i = kept + tid; i < maxdeg_p2; i += blockDim.x)
            newk[i] = ~0ull;
        __syncthreads();

        /* bitonic sort of the <=maxdeg_p2 new keys (mask out flag bits) */
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

        /* merge-path: C = first min(sz+kept, efc) of merge(list, newk) */
        const int outn = min(sz + kept, efc);
        for (int i = tid; i < outn; i += blockDim.x) {
            int lo = max(0, i - kept), hi = min(i, sz);
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
            const uint64_t vb = b < kept ? newk[b] : ~0ull;
            merged[i] = (va & ~EXPANDED_BIT) <= (vb & ~EXPANDED_BIT) ? va : vb;
        }
        __syncthreads();
        for (int i = tid; i < outn; i += blockDim.x) list[i] = merged[i];
        if (tid == 0) sc[0] = outn;
        __syncthreads();
    }

    /* output first k (ascending, PASSING only when FILTERED), padded */
    const int sz = sc[0];
    if (FILTERED) {
        /* order-preserving compaction of passing entries (thread 0: <=efc
         * scan once per query) */
        if (tid == 0) {
            int o = 0;
            for (int i = 0; i < sz && o < k; ++i)
                if (list[i] & PASS_BIT) merged[o++] = list[i];
            sc[1] = o;
        }
        __syncthreads();
        const int o = sc[1];
        for (int i = tid; i < k; i += blockDim.x) {
            if (i < o) {
                const uint64_t kv = merged[i];
                out_ids[(int64_t)q * k + i] = key_of_slot[kv & slot_mask];
                out_dists[(int64_t)q * k + i] = hu2f((uint32_t)(kv >> 32));
            } else {
                out_ids[(int64_t)q * k + i] = -1;
                out_dists[(int64_t)q * k + i] = FLT_MAX;
            }
        }
    } else
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
                        int ef, int k, int maxdeg_p2,
                        const uint32_t* slot_filter, uint32_t* visited,
                        int64_t vwords, const int64_t* key_of_slot,
                        int64_t* out_ids, float* out_dists,
                        unsigned long long* eval_count, size_t* shmem_out,
                        hipStream_t stream) {
    const bool filtered = slot_filter != nullptr;
    const int efc = filtered ? 2 * ef : ef;
    const size_t shmem = ((dpad * 4 + 7) & ~7) + (size_t)efc * 8 * 2 +
                         (size_t)maxdeg_p2 * 8 + (size_t)maxdeg_p2 * 4 +
                         16 * 4 + 8 * 4 + 1024 * 4 + 16;
    if (shmem_out) *shmem_out = shmem;
    if (!nq) return;
#define HARGS vecs, dpad, n, l0_offs, l0_nbrs, u_idx, u_count, up_offs,     \
    up_nbrs, max_level, entry_slot, queries, qnorms, ef, efc, k, maxdeg_p2, \
    slot_filter, visited, vwords, key_of_slot, out_ids, out_dists, eval_count
#define HLAUNCH(M, F)                                                       \
    hipLaunchKernelGGL((hnsw_search_kernel<M, F>), dim3(nq), dim3(1024),    \
                       shmem, stream, HARGS)
    if (filtered) {
        switch (metric) {
        case U_L2SQ: HLAUNCH(U_L2SQ, true); break;
        case U_IP: HLAUNCH(U_IP, true); break;
        default: HLAUNCH(U_COS, true); break;
        }
    } else {
        switch (metric) {
        case U_L2SQ: HLAUNCH(U_L2SQ, false); break;
        case U_IP: HLAUNCH(U_IP, false); break;
        default: HLAUNCH(U_COS, false); break;
        }
    }
#undef HLAUNCH
#undef HARGS
}

}  // namespace moann
