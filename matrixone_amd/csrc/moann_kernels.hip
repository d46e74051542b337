/* CDNA4 (gfx950) device kernels for the MatrixOne ANN search hot path.
 *
 * Written MI355X-first (not a port): 64-wide wavefronts, coalesced
 * float4 streams over an interleaved HBM layout, LDS-staged query tiles,
 * per-query radix select. The list-scan kernel replaces the reference's
 * per-row SimSIMD distance loop (pkg/vectorindex/ivfflat/relation_search.go:
 * 334-368 + pkg/sql/plan/function/func_binary.go:9045-9175) and the cuVS
 * ivf_flat list-scan launch (cgo/cuvs/ivf_flat.hpp:921-945 precedent);
 * results follow MO's distance conventions (metric/distance_func.go:
 * IP = -x.q at :207, cosine clamp + denom-0 at :211-286).
 *
 * Data layout: vectors are packed per list into groups of 64 rows,
 * dim-major inside the group in float4 quads:
 *   packed[(g*dpad + 4*j4)*64 + lane*4 + c] = row(g*64+lane) dim(4*j4+c)
 * so one wave reading quad j4 for its 64 rows issues one fully coalesced
 * 1 KiB load (16 B/lane). Queries are tiled (QT<=8 per workgroup) in LDS and
 * broadcast-read, so each HBM byte of list data serves QT queries.
 */

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <algorithm>
#include <cfloat>
#include <cstdint>
#include <cstdlib>
#include <cstring>

#include "moann_internal.h"

namespace moann {

#define WAVE 64

/* monotone float -> u32 key (ascending order preserved) */
__device__ __forceinline__ uint32_t f2u(float f) {
    uint32_t u = __float_as_uint(f);
    return u ^ (((int32_t)u >> 31) | 0x80000000u);
}
__device__ __forceinline__ float u2f(uint32_t u) {
    u ^= (((int32_t)(u ^ 0x80000000u)) >> 31) | 0x80000000u;
    return __uint_as_float(u);
}

/* ------------------------------ scan -------------------------------------
 * One workgroup (256 threads = 4 waves) per job. Each wave walks a pair of
 * adjacent 64-row groups so every LDS broadcast of a query quad feeds two
 * 16 B data quads (halves LDS traffic per byte). Per j4 step and query:
 * 8 fma-class VALU ops per 32 B/lane -> VALU ceiling ~19 TB/s equivalent,
 * comfortably above the ~6.3 TB/s HBM bound this kernel targets. */
template <int METRIC, int QT, int GP>
__global__ __launch_bounds__(256) void scan_kernel(
    const float* __restrict__ packed, const float* __restrict__ queries,
    const float* __restrict__ qnorms, int dpad,
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gstart, const int32_t* __restrict__ j_gcount,
    const int32_t* __restrict__ j_rows, const int32_t* __restrict__ j_nq,
    const int32_t* __restrict__ j_qbase,
    const int32_t* __restrict__ qslot_query,
    const int64_t* __restrict__ qslot_outbase,
    const int64_t* __restrict__ j_slot_base,
    const uint32_t* __restrict__ filter_bitset, /* bit per GLOBAL slot; a
        cleared bit excludes the entry (cuVS bitset_filter precedent,
        cgo/cuvs/ivf_flat.hpp:908-924); null = unfiltered */
    float* __restrict__ dists_out) {
    extern __shared__ float lds[];  /* [QT][dpad] (+[QT] qnorm for cos) */
    float* ldsq = lds;
    float* ldsn = lds + QT * dpad;

    /* XCD-aware bijective blockIdx remap (T1): the dispatcher places block
     * b on XCD b%8; this permutation gives each XCD a CONTIGUOUS job range,
     * so the query tiles of one list (consecutive jobs) stream through one
     * XCD's L2 instead of re-fetching from HBM on every XCD. */
    int j;
    {
        const int nwg = gridDim.x, bid = blockIdx.x;
        const int q8 = nwg >> 3, r8 = nwg & 7;
        const int xcd = bid & 7, idx = bid >> 3;
        j = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
        if (j >= nwg) j = bid; /* safety: cannot happen for bijective map */
    }
    const int nq = j_nq[j];
    const int qbase = j_qbase[j];

    for (int t = 0; t < QT; ++t) {
        if (t < nq) {
            const int q = qslot_query[qbase + t];
            const float* src = queries + (int64_t)q * dpad;
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = src[e];
            if (METRIC == KM_COS && threadIdx.x == 0) ldsn[t] = qnorms[q];
        } else {
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = 0.f;
            if (METRIC == KM_COS && threadIdx.x == 0) ldsn[t] = 0.f;
        }
    }
    __syncthreads();

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gstart = j_gstart[j], gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];
    const int d4 = dpad >> 2;

    /* GP=2: each wave walks ADJACENT group pairs (one LDS broadcast feeds
     * 32 B of data — best for QT<=8). GP=1: one group per wave — half the
     * live registers, which is what lets hipcc software-pipeline the loads
     * at QT=16. */
    for (int gp = gstart + GP * wave; gp < gstart + gcount; gp += 4 * GP) {
        const int g0 = gp;
        const bool has1 = GP == 2 && (gp + 1) < (gstart + gcount);
        const int g1 = has1 ? gp + 1 : gp;
        const float4* __restrict__ d0 =
            (const float4*)(packed + (baseg + g0) * (int64_t)64 * dpad) + lane;
        const float4* __restrict__ d1 =
            (const float4*)(packed + (baseg + g1) * (int64_t)64 * dpad) + lane;

        float acc0[QT], acc1[QT];
#pragma unroll
        for (int t = 0; t < QT; ++t) acc0[t] = acc1[t] = 0.f;
        float rn0 = 0.f, rn1 = 0.f;

        /* GP==1: hand-rolled 3-deep load ring in inline asm. hipcc will not
         * software-pipeline this loop at QT=16 register pressure (it emits a
         * full vmcnt(0) drain per iteration); the asm ring keeps 2 loads in
         * flight across iterations with counted waits (guide rules: asm
         * loads are invisible to hipcc's s_waitcnt bookkeeping, so the
         * counted wait + sched_barrier(0) before the first consumer is
         * ours). */
        if (GP == 1) {
            /* named ring slots (no arrays: a runtime-indexed ext_vector
             * array goes to scratch) */
            float4 r0, r1, r2;
#define MOANN_ISSUE(slot, idx)                                              \
            {                                                               \
                const float4* ap = d0 + ((idx) < d4 ? (idx) : d4 - 1) * 64; \
                asm volatile("global_load_dwordx4 %0, %1, off"              \
                             : "=v"(slot) : "v"(ap));                       \
            }
#define MOANN_CONSUME(slot, idx)                                            \
            {                                                               \
                asm volatile("s_waitcnt vmcnt(2)" ::: "memory");            \
                /* full fence (mask 0) is REQUIRED: any relaxation that     \
                 * lets VMEM ops cross lets a ring REISSUE hoist above this \
                 * wait, the outstanding count then under-retires and the   \
                 * consumer reads an un-landed slot (measured: recall 0.0   \
                 * at mask DS_READ|VMEM_READ while 1.6x "faster").          \
                 * Counted-wait discipline = nothing crosses the wait.      */\
                __builtin_amdgcn_sched_barrier(0);                          \
                const float4 x0 = slot;                                     \
                const int q4i = (idx);                                      \
                if (METRIC == KM_COS)                                       \
                    rn0 += x0.x * x0.x + x0.y * x0.y + x0.z * x0.z +        \
                           x0.w * x0.w;                                     \
                _Pragma("unroll")                                           \
                for (int t = 0; t < QT; ++t) {                              \
                    const float4 qv =                                       \
                        ((const float4*)(ldsq + t * dpad))[q4i];             \
                    if (METRIC == KM_L2SQ) {                                \
                        float e;                                            \
                        e = x0.x - qv.x; acc0[t] = fmaf(e, e, acc0[t]);     \
                        e = x0.y - qv.y; acc0[t] = fmaf(e, e, acc0[t]);     \
                        e = x0.z - qv.z; acc0[t] = fmaf(e, e, acc0[t]);     \
                        e = x0.w - qv.w; acc0[t] = fmaf(e, e, acc0[t]);     \
                    } else if (METRIC == KM_IP || METRIC == KM_COS) {       \
                        acc0[t] = fmaf(x0.x, qv.x, acc0[t]);                \
                        acc0[t] = fmaf(x0.y, qv.y, acc0[t]);                \
                        acc0[t] = fmaf(x0.z, qv.z, acc0[t]);                \
                        acc0[t] = fmaf(x0.w, qv.w, acc0[t]);                \
                    } else {                                                \
                        acc0[t] += fabsf(x0.x - qv.x) + fabsf(x0.y - qv.y) +\
                                   fabsf(x0.z - qv.z) + fabsf(x0.w - qv.w); \
                    }                                                       \
                }                                                           \
            }
            MOANN_ISSUE(r0, 0)
            MOANN_ISSUE(r1, 1)
            MOANN_ISSUE(r2, 2)
            int q4 = 0;
            for (; q4 + 3 <= d4; q4 += 3) {
                MOANN_CONSUME(r0, q4)
                MOANN_ISSUE(r0, q4 + 3)
                MOANN_CONSUME(r1, q4 + 1)
                MOANN_ISSUE(r1, q4 + 4)
                MOANN_CONSUME(r2, q4 + 2)
                MOANN_ISSUE(r2, q4 + 5)
            }
            /* tail (<=2 iterations): drain, then consume without reissue */
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_sched_barrier(0);
            if (q4 < d4) { MOANN_CONSUME(r0, q4) }
            if (q4 + 1 < d4) { MOANN_CONSUME(r1, q4 + 1) }
#undef MOANN_ISSUE
#undef MOANN_CONSUME
        } else
#pragma unroll 2 /* two iterations in flight: 4 dwordx4 issued together,
                    counted vmcnt waits (hipcc pipelines the pair the same
                    way it does for the QT=8 instantiation) */
        for (int q4 = 0; q4 < d4; ++q4) {
            const float4 x0 = d0[q4 * 64];
            const float4 x1 = GP == 2 ? d1[q4 * 64] : float4 {0, 0, 0, 0};
            if (METRIC == KM_COS) {
                rn0 += x0.x * x0.x + x0.y * x0.y + x0.z * x0.z + x0.w * x0.w;
                if (GP == 2)
                    rn1 += x1.x * x1.x + x1.y * x1.y + x1.z * x1.z + x1.w * x1.w;
            }
#pragma unroll
            for (int t = 0; t < QT; ++t) {
                const float4 qv = ((const float4*)(ldsq + t * dpad))[q4];
                if (METRIC == KM_L2SQ) {
                    float e;
                    e = x0.x - qv.x; acc0[t] = fmaf(e, e, acc0[t]);
                    e = x0.y - qv.y; acc0[t] = fmaf(e, e, acc0[t]);
                    e = x0.z - qv.z; acc0[t] = fmaf(e, e, acc0[t]);
                    e = x0.w - qv.w; acc0[t] = fmaf(e, e, acc0[t]);
                    if (GP == 2) {
                        e = x1.x - qv.x; acc1[t] = fmaf(e, e, acc1[t]);
                        e = x1.y - qv.y; acc1[t] = fmaf(e, e, acc1[t]);
                        e = x1.z - qv.z; acc1[t] = fmaf(e, e, acc1[t]);
                        e = x1.w - qv.w; acc1[t] = fmaf(e, e, acc1[t]);
                    }
                } else if (METRIC == KM_IP || METRIC == KM_COS) {
                    acc0[t] = fmaf(x0.x, qv.x, acc0[t]);
                    acc0[t] = fmaf(x0.y, qv.y, acc0[t]);
                    acc0[t] = fmaf(x0.z, qv.z, acc0[t]);
                    acc0[t] = fmaf(x0.w, qv.w, acc0[t]);
                    if (GP == 2) {
                        acc1[t] = fmaf(x1.x, qv.x, acc1[t]);
                        acc1[t] = fmaf(x1.y, qv.y, acc1[t]);
                        acc1[t] = fmaf(x1.z, qv.z, acc1[t]);
                        acc1[t] = fmaf(x1.w, qv.w, acc1[t]);
                    }
                } else { /* KM_L1 */
                    acc0[t] += fabsf(x0.x - qv.x) + fabsf(x0.y - qv.y) +
                               fabsf(x0.z - qv.z) + fabsf(x0.w - qv.w);
                    if (GP == 2)
                        acc1[t] += fabsf(x1.x - qv.x) + fabsf(x1.y - qv.y) +
                                   fabsf(x1.z - qv.z) + fabsf(x1.w - qv.w);
                }
            }
        }

        const int row0 = g0 * 64 + lane;
        const int row1 = g1 * 64 + lane;
        bool pass0 = true, pass1 = true;
        if (filter_bitset) {
            const int64_t sb = j_slot_base[j];
            if (row0 < rows) {
                const int64_t gs = sb + row0;
                pass0 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
            if (has1 && row1 < rows) {
                const int64_t gs = sb + row1;
                pass1 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
        }
#pragma unroll
        for (int t = 0; t < QT; ++t) {
            if (t >= nq) break;
            const int64_t ob = qslot_outbase[qbase + t];
            if (row0 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = -acc0[t];
                else if (METRIC == KM_COS) {
                    /* distance_func.go:211-286: double denom, clamp, 0 -> 1 */
                    double denom = sqrt((double)rn0) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc0[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else dist = acc0[t];
                dists_out[ob + row0] = pass0 ? dist : FLT_MAX;
            }
            if (has1 && row1 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = -acc1[t];
                else if (METRIC == KM_COS) {
                    double denom = sqrt((double)rn1) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc1[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else dist = acc1[t];
                dists_out[ob + row1] = pass1 ? dist : FLT_MAX;
            }
        }
    }
}

template <int METRIC>
static void launch_scan_qt(int qt, const float* packed, const float* queries,
                           const float* qnorms, int dpad, const ScanJobs& jb,
                           float* dists_out, hipStream_t stream,
                           const uint32_t* filter_bitset) {
    const dim3 grid(jb.njobs), block(256);
#define ARGS packed, queries, qnorms, dpad, jb.databaseg, jb.gstart, \
    jb.gcount, jb.rows, jb.nq, jb.qbase, jb.qslot_query, jb.qslot_outbase, \
    jb.slot_base, filter_bitset, dists_out
    switch (qt) {
    case 16:
        hipLaunchKernelGGL((scan_kernel<METRIC, 16, 1>), grid, block,
                           (16 * dpad + 16) * sizeof(float), stream, ARGS);
        break;
    case 12:
        hipLaunchKernelGGL((scan_kernel<METRIC, 12, 2>), grid, block,
                           (12 * dpad + 12) * sizeof(float), stream, ARGS);
        break;
    case 8:
        hipLaunchKernelGGL((scan_kernel<METRIC, 8, 2>), grid, block,
                           (8 * dpad + 8) * sizeof(float), stream, ARGS);
        break;
    case 4:
        hipLaunchKernelGGL((scan_kernel<METRIC, 4, 2>), grid, block,
                           (4 * dpad + 4) * sizeof(float), stream, ARGS);
        break;
    case 2:
        hipLaunchKernelGGL((scan_kernel<METRIC, 2, 2>), grid, block,
                           (2 * dpad + 2) * sizeof(float), stream, ARGS);
        break;
    default:
        hipLaunchKernelGGL((scan_kernel<METRIC, 1, 2>), grid, block,
                           (1 * dpad + 1) * sizeof(float), stream, ARGS);
        break;
    }
#undef ARGS
}

void launch_scan(int metric, int qt, const float* packed, const float* queries,
                 const float* qnorms, int dpad, const ScanJobs& jb,
                 float* dists_out, hipStream_t stream,
                 const uint32_t* filter_bitset) {
    if (jb.njobs == 0) return;
    /* flagship shape -> hand-scheduled asm kernel (scan_asm768.hip);
     * MOANN_SCAN=generic forces the compiler-scheduled path for A/B */
    if (metric == KM_L2SQ && qt == 16 && dpad == 768) {
        static const bool generic = [] {
            const char* e = getenv("MOANN_SCAN");
            return e && strcmp(e, "generic") == 0;
        }();
        if (!generic) {
            launch_scan_asm768(packed, queries, dpad, jb, dists_out, stream,
                               filter_bitset);
            return;
        }
    }
    switch (metric) {
    case KM_L2SQ: launch_scan_qt<KM_L2SQ>(qt, packed, queries, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    case KM_IP:   launch_scan_qt<KM_IP>(qt, packed, queries, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    case KM_COS:  launch_scan_qt<KM_COS>(qt, packed, queries, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    default:      launch_scan_qt<KM_L1>(qt, packed, queries, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    }
}

/* ------------------------------ top-k ------------------------------------
 * One workgroup per query: 4-level 8-bit radix select over the monotone u32
 * keys (exact kth threshold), then gather of the k winners and an in-LDS
 * bitonic sort of packed (key32|slot32) u64s — ascending by distance, slot as
 * the tie-breaker, giving deterministic output. k <= 4096.
 * Replaces the bounded heaps of pkg/vectorindex/index.go:135-309 and the
 * cuVS select::topk stage. Padding: (-1, FLT_MAX) per cgo/cuvs/helper.h:160. */
__global__ __launch_bounds__(256) void topk_kernel(
    const float* __restrict__ dists, const int64_t* __restrict__ qoffs,
    int64_t uniform_n, int k, int sortP,
    int32_t* __restrict__ out_slots, float* __restrict__ out_dists) {
    extern __shared__ unsigned char smem[];
    uint32_t* hist = (uint32_t*)smem;                    /* 256 */
    uint32_t* ctr = hist + 256;                          /* n_lt, n_tie */
    uint32_t* st = ctr + 8;  /* prefix, c_sure, tieval, need_ties, done */
    uint64_t* keys = (uint64_t*)(smem + 2048);           /* [sortP] */

    const int q = blockIdx.x;
    const int64_t off = qoffs ? qoffs[q] : (int64_t)q * uniform_n;
    const int64_t count = (qoffs ? qoffs[q + 1] : (int64_t)(q + 1) * uniform_n) - off;
    const int tid = threadIdx.x;
    const int m = (int)(count < (int64_t)k ? count : k);
    int nkeys;

    if (count > (int64_t)k) {
        /* radix select for the exact threshold. Each full-array read is
         * 4x unrolled so every wave keeps 4 loads in flight (a single
         * strided scalar load per wave is latency-bound at ~1.3 TB/s;
         * measured r01_fetch_10M_asm_cg8.csv). */
        if (tid == 0) { st[0] = 0; st[1] = 0; st[3] = 0; st[4] = 0; st[5] = 0; }
        __syncthreads();
        for (int level = 0; level < 4 && !st[4]; ++level) {
            const int shift = 24 - 8 * level;
            for (int b = tid; b < 256; b += blockDim.x) hist[b] = 0;
            __syncthreads();
            const uint32_t pref = st[0];
            const int64_t step = blockDim.x;
            int64_t i = tid;
#define MOANN_HISTO(u)                                                                  if (level == 0 || ((u) >> (shift + 8)) == (pref >> (shift + 8)))                atomicAdd(&hist[((u) >> shift) & 255], 1u);
            for (; i + 3 * step < count; i += 4 * step) {
                const uint32_t u0 = f2u(dists[off + i]);
                const uint32_t u1 = f2u(dists[off + i + step]);
                const uint32_t u2 = f2u(dists[off + i + 2 * step]);
                const uint32_t u3 = f2u(dists[off + i + 3 * step]);
                MOANN_HISTO(u0) MOANN_HISTO(u1) MOANN_HISTO(u2) MOANN_HISTO(u3)
            }
            for (; i < count; i += step) {
                const uint32_t u = f2u(dists[off + i]);
                MOANN_HISTO(u)
            }
#undef MOANN_HISTO
            __syncthreads();
            if (tid == 0) {
                const uint32_t need = (uint32_t)k - st[1];
                uint32_t cum = 0; int b = 0;
                for (; b < 256; ++b) {
                    if (cum + hist[b] >= need) break;
                    cum += hist[b];
                }
                st[0] = pref | ((uint32_t)b << shift);
                st[1] += cum;
                if (cum + hist[b] == need) {
                    /* whole bin selected: no tie resolution needed */
                    st[1] += hist[b];
                    st[0] |= (shift ? ((1u << shift) - 1u) : 0u);
                    st[3] = 0; st[4] = 1;
                } else if (st[1] + hist[b] <= (uint32_t)sortP) {
                    /* boundary GROUP fits in the sort buffer: collect the
                     * whole prefix group and let the bitonic sort truncate.
                     * Exact (every candidate strictly below the group is
                     * kept) and deterministic (full keys are distinct), and
                     * it usually ends the scan at level 0 or 1 instead of
                     * re-reading the array for levels 2-3. */
                    st[2] = st[0];
                    st[3] = hist[b];
                    st[5] = (uint32_t)shift;
                    st[4] = 1;
                } else if (level == 3) {
                    st[2] = st[0];
                    st[3] = need - cum;
                    st[5] = 0;
                }
            }
            __syncthreads();
        }
        const uint32_t c_sure = st[1];
        const uint32_t tieval = st[2];
        const uint32_t need_ties = st[3];
        const uint32_t tie_shift = st[5];
        const int exact = st[4] && !need_ties;
        /* strict-below bound: exact-exit keeps everything <= prefix (low
         * bits already forced to 1); group/tie modes keep u < prefix (low
         * bits zero, so < excludes the whole boundary group) */
        const uint64_t below = (uint64_t)st[0] + (exact ? 1u : 0u);
        if (tid == 0) { ctr[0] = 0; ctr[1] = 0; }
        __syncthreads();
        {
            const int64_t step = blockDim.x;
            int64_t i = tid;
#define MOANN_COLLECT(u, idx)                                                           if ((uint64_t)(u) < below) {                                                    const uint32_t pos = atomicAdd(&ctr[0], 1u);                                keys[pos] = ((uint64_t)(u) << 32) | (uint32_t)(idx);                    } else if (need_ties &&                                                                ((u) >> tie_shift) == (tieval >> tie_shift)) {                       const uint32_t t = atomicAdd(&ctr[1], 1u);                                  if (t < need_ties)                                                              keys[c_sure + t] = ((uint64_t)(u) << 32) | (uint32_t)(idx);             }
            for (; i + 3 * step < count; i += 4 * step) {
                const uint32_t u0 = f2u(dists[off + i]);
                const uint32_t u1 = f2u(dists[off + i + step]);
                const uint32_t u2 = f2u(dists[off + i + 2 * step]);
                const uint32_t u3 = f2u(dists[off + i + 3 * step]);
                MOANN_COLLECT(u0, i) MOANN_COLLECT(u1, i + step)
                MOANN_COLLECT(u2, i + 2 * step) MOANN_COLLECT(u3, i + 3 * step)
            }
            for (; i < count; i += step) {
                const uint32_t u = f2u(dists[off + i]);
                MOANN_COLLECT(u, i)
            }
#undef MOANN_COLLECT
        }
        __syncthreads();
        nkeys = (int)(c_sure + need_ties);
    } else {
        for (int64_t i = tid; i < count; i += blockDim.x)
            keys[i] = ((uint64_t)f2u(dists[off + i]) << 32) | (uint32_t)i;
        __syncthreads();
        nkeys = m;
    }

    /* pad from the number of COLLECTED keys (group-collect gathers up to
     * sortP entries, more than k) and let the sort truncate to k */
    for (int i = nkeys + tid; i < sortP; i += blockDim.x) keys[i] = ~0ULL;
    __syncthreads();

    /* bitonic sort ascending over sortP u64 keys */
    for (int ks = 2; ks <= sortP; ks <<= 1) {
        for (int jj = ks >> 1; jj > 0; jj >>= 1) {
            for (int i = tid; i < sortP; i += blockDim.x) {
                const int l = i ^ jj;
                if (l > i) {
                    const uint64_t a = keys[i], b = keys[l];
                    const bool up = (i & ks) == 0;
                    if ((a > b) == up) { keys[i] = b; keys[l] = a; }
                }
            }
            __syncthreads();
        }
    }

    for (int i = tid; i < k; i += blockDim.x) {
        if (i < m) {
            const uint64_t kv = keys[i];
            out_slots[(int64_t)q * k + i] = (int32_t)(kv & 0xFFFFFFFFu);
            out_dists[(int64_t)q * k + i] = u2f((uint32_t)(kv >> 32));
        } else {
            out_slots[(int64_t)q * k + i] = -1;
            out_dists[(int64_t)q * k + i] = FLT_MAX;
        }
    }
}

void launch_topk(const float* dists, const int64_t* qoffs, int64_t uniform_n,
                 int nq, int k, int32_t* out_slots, float* out_dists,
                 hipStream_t stream) {
    if (nq == 0) return;
    int sortP = 1;
    while (sortP < k) sortP <<= 1;
    /* A larger sort buffer makes the group-collect exit fire at level 0
     * (strict-below <= k plus a ~count/256 boundary group): 2 full-array
     * reads instead of 3 for the big per-query candidate selections. The
     * extra bitonic work (1024 u64 keys) is microseconds per block; LDS
     * 2048+8*1024 = 10 KB still keeps several blocks per CU. */
    if (sortP < 1024) sortP = 1024;
    const size_t shmem = 2048 + (size_t)sortP * 8;
    hipLaunchKernelGGL(topk_kernel, dim3(nq), dim3(256), shmem, stream,
                       dists, qoffs, uniform_n, k, sortP, out_slots, out_dists);
}

/* ------------------------------ gather ----------------------------------- */

__device__ __forceinline__ float transform_score_dev(float raw, int do_sqrt,
                                                     double inv_mul2) {
    /* ivfflat/search.go:1062-1077 + metric/types.go:245-251 */
    double r = (double)raw * inv_mul2;
    if (do_sqrt) r = sqrt(r);
    return (float)r;
}

__global__ void gather_kernel(
    const int32_t* __restrict__ sel_slots, const float* __restrict__ sel_dists,
    const int32_t* __restrict__ probe_lists, const int64_t* __restrict__ probe_offs,
    const int64_t* __restrict__ list_slot_base, const int64_t* __restrict__ id_by_slot,
    int nprobe, int nq, int k, int do_sqrt, double inv_mul2,
    int64_t* __restrict__ out_ids, float* __restrict__ out_dists) {
    const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)nq * k) return;
    const int q = (int)(idx / k);
    const int32_t slot = sel_slots[idx];
    /* FLT_MAX marks padding AND membership-filtered rows */
    if (slot < 0 || sel_dists[idx] == FLT_MAX) {
        out_ids[idx] = -1; out_dists[idx] = FLT_MAX; return;
    }
    /* binary search rank r: probe_offs[q][r] <= slot < probe_offs[q][r+1] */
    const int64_t* offs = probe_offs + (int64_t)q * (nprobe + 1);
    int lo = 0, hi = nprobe - 1;
    while (lo < hi) {
        const int mid = (lo + hi + 1) >> 1;
        if (offs[mid] <= slot) lo = mid; else hi = mid - 1;
    }
    const int32_t list = probe_lists[(int64_t)q * nprobe + lo];
    const int64_t local = slot - offs[lo];
    out_ids[idx] = id_by_slot[list_slot_base[list] + local];
    out_dists[idx] = transform_score_dev(sel_dists[idx], do_sqrt, inv_mul2);
}

void launch_gather(const int32_t* sel_slots, const float* sel_dists,
                   const int32_t* probe_lists, const int64_t* probe_offs,
                   const int64_t* list_slot_base, const int64_t* id_by_slot,
                   int nprobe, int nq, int k, int do_sqrt, double inv_mul2,
                   int64_t* out_ids, float* out_dists, hipStream_t stream) {
    const int64_t total = (int64_t)nq * k;
    if (!total) return;
    const int block = 256;
    const int grid = (int)((total + block - 1) / block);
    hipLaunchKernelGGL(gather_kernel, dim3(grid), dim3(block), 0, stream,
                       sel_slots, sel_dists, probe_lists, probe_offs,
                       list_slot_base, id_by_slot, nprobe, nq, k, do_sqrt,
                       inv_mul2, out_ids, out_dists);
}

__global__ void gather_uniform_kernel(
    const int32_t* __restrict__ sel_slots, const float* __restrict__ sel_dists,
    const int64_t* __restrict__ id_by_slot, int nq, int k, int do_sqrt,
    double inv_mul2, int64_t* __restrict__ out_ids,
    float* __restrict__ out_dists) {
    const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)nq * k) return;
    const int32_t slot = sel_slots[idx];
    if (slot < 0) { out_ids[idx] = -1; out_dists[idx] = FLT_MAX; return; }
    out_ids[idx] = id_by_slot ? id_by_slot[slot] : (int64_t)slot;
    out_dists[idx] = transform_score_dev(sel_dists[idx], do_sqrt, inv_mul2);
}

void launch_gather_uniform(const int32_t* sel_slots, const float* sel_dists,
                           const int64_t* id_by_slot, int nq, int k,
                           int do_sqrt, double inv_mul2, int64_t* out_ids,
                           float* out_dists, hipStream_t stream) {
    const int64_t total = (int64_t)nq * k;
    if (!total) return;
    const int block = 256;
    const int grid = (int)((total + block - 1) / block);
    hipLaunchKernelGGL(gather_uniform_kernel, dim3(grid), dim3(block), 0,
                       stream, sel_slots, sel_dists, id_by_slot, nq, k,
                       do_sqrt, inv_mul2, out_ids, out_dists);
}

/* ------------------------------ pack -------------------------------------
 * One-time permutation into the interleaved layout: thread = one output
 * float4; writes coalesced, reads gather (4 B x 4 from the source row). */
__global__ void pack_kernel(const float* __restrict__ vecs, int dim, int dpad,
                            const int64_t* __restrict__ group_rowbase,
                            const int32_t* __restrict__ group_valid,
                            const int64_t* __restrict__ slot_rows,
                            int64_t ngroups, float* __restrict__ packed) {
    const int64_t nquads = ngroups * (dpad >> 2) * 64;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < nquads; idx += stride) {
        const int lane = (int)(idx & 63);
        const int64_t rest = idx >> 6;
        const int j4 = (int)(rest % (dpad >> 2));
        const int64_t g = rest / (dpad >> 2);
        float4 v = {0.f, 0.f, 0.f, 0.f};
        if (lane < group_valid[g]) {
            const int64_t row = slot_rows[group_rowbase[g] + lane];
            const float* src = vecs + row * (int64_t)dim;
            const int base = 4 * j4;
            v.x = base + 0 < dim ? src[base + 0] : 0.f;
            v.y = base + 1 < dim ? src[base + 1] : 0.f;
            v.z = base + 2 < dim ? src[base + 2] : 0.f;
            v.w = base + 3 < dim ? src[base + 3] : 0.f;
        }
        ((float4*)packed)[idx] = v;
    }
}

void launch_pack(const float* vecs, int dim, int dpad,
                 const int64_t* group_rowbase, const int32_t* group_valid,
                 const int64_t* slot_rows, int64_t ngroups, float* packed,
                 hipStream_t stream) {
    const int64_t nquads = ngroups * (int64_t)(dpad >> 2) * 64;
    if (!nquads) return;
    const int block = 256;
    const int64_t grid = std::min<int64_t>((nquads + block - 1) / block,
                                           1 << 22);
    hipLaunchKernelGGL(pack_kernel, dim3((uint32_t)grid), dim3(block), 0,
                       stream, vecs, dim, dpad, group_rowbase, group_valid,
                       slot_rows, ngroups, packed);
}

/* ------------------------------ unpack -----------------------------------
 * Inverse of pack_kernel for save_dir: packed groups -> row-major f32 rows
 * in SLOT order (valid slots only), one chunk of slots at a time. */
__global__ void unpack_kernel(const float* __restrict__ packed, int dim,
                              int dpad,
                              const int64_t* __restrict__ group_slotbase,
                              const int32_t* __restrict__ group_valid,
                              int64_t ngroups, int64_t slot_lo,
                              int64_t slot_hi, float* __restrict__ out) {
    const int64_t nquads = ngroups * (dpad >> 2) * 64;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < nquads; idx += stride) {
        const int lane = (int)(idx & 63);
        const int64_t rest = idx >> 6;
        const int j4 = (int)(rest % (dpad >> 2));
        const int64_t g = rest / (dpad >> 2);
        if (lane >= group_valid[g]) continue;
        const int64_t slot = group_slotbase[g] + lane;
        if (slot < slot_lo || slot >= slot_hi) continue;
        const float4 v = ((const float4*)packed)[idx];
        float* dst = out + (slot - slot_lo) * (int64_t)dim;
        const int base = 4 * j4;
        if (base + 0 < dim) dst[base + 0] = v.x;
        if (base + 1 < dim) dst[base + 1] = v.y;
        if (base + 2 < dim) dst[base + 2] = v.z;
        if (base + 3 < dim) dst[base + 3] = v.w;
    }
}

void launch_unpack(const float* packed, int dim, int dpad,
                   const int64_t* group_slotbase, const int32_t* group_valid,
                   int64_t ngroups, int64_t slot_lo, int64_t slot_hi,
                   float* out, hipStream_t stream) {
    const int64_t nquads = ngroups * (int64_t)(dpad >> 2) * 64;
    if (!nquads) return;
    const int block = 256;
    const int64_t grid = std::min<int64_t>((nquads + block - 1) / block,
                                           1 << 22);
    hipLaunchKernelGGL(unpack_kernel, dim3((uint32_t)grid), dim3(block), 0,
                       stream, packed, dim, dpad, group_slotbase, group_valid,
                       ngroups, slot_lo, slot_hi, out);
}

/* byte-storage inverse of pack: packed bytes -> row-major [slot][dim]
 * bytes, both layouts (layout16 = [g][dpad/16][64][16] when dpad%16==0,
 * else the uchar4 [g][dpad/4][64][4]). save_dir for narrow storage. */
template <bool L16>
__global__ void unpack_bytes_kernel(const uint8_t* __restrict__ packed,
                                    int dim, int dpad,
                                    const int64_t* __restrict__ group_slotbase,
                                    const int32_t* __restrict__ group_valid,
                                    int64_t ngroups, int64_t slot_lo,
                                    int64_t slot_hi,
                                    uint8_t* __restrict__ out) {
    const int U = L16 ? 16 : 4;
    const int64_t nunits = ngroups * (int64_t)(dpad / U) * 64;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < nunits; idx += stride) {
        const int lane = (int)(idx & 63);
        const int64_t rest = idx >> 6;
        const int m = (int)(rest % (dpad / U));
        const int64_t g = rest / (dpad / U);
        if (lane >= group_valid[g]) continue;
        const int64_t slot = group_slotbase[g] + lane;
        if (slot < slot_lo || slot >= slot_hi) continue;
        const uint8_t* src = packed + idx * U;
        uint8_t* dst = out + (slot - slot_lo) * (int64_t)dim;
        const int base = m * U;
#pragma unroll
        for (int b = 0; b < U; ++b)
            if (base + b < dim) dst[base + b] = src[b];
    }
}

void launch_unpack_bytes(const uint8_t* packed, int dim, int dpad,
                         const int64_t* group_slotbase,
                         const int32_t* group_valid, int64_t ngroups,
                         int64_t slot_lo, int64_t slot_hi, uint8_t* out,
                         hipStream_t stream) {
    const bool l16 = (dpad & 15) == 0;
    const int U = l16 ? 16 : 4;
    const int64_t nunits = ngroups * (int64_t)(dpad / U) * 64;
    if (!nunits) return;
    const int block = 256;
    const int64_t grid = std::min<int64_t>((nunits + block - 1) / block,
                                           1 << 22);
    if (l16)
        hipLaunchKernelGGL(unpack_bytes_kernel<true>, dim3((uint32_t)grid),
                           dim3(block), 0, stream, packed, dim, dpad,
                           group_slotbase, group_valid, ngroups, slot_lo,
                           slot_hi, out);
    else
        hipLaunchKernelGGL(unpack_bytes_kernel<false>, dim3((uint32_t)grid),
                           dim3(block), 0, stream, packed, dim, dpad,
                           group_slotbase, group_valid, ngroups, slot_lo,
                           slot_hi, out);
}

/* ---------------- half-precision (f16 / bf16) storage -------------------
 * MO's float16/bf16 quantization is a PLAIN CAST (quantizer.go:50-58,
 * CastSQL:179-183): entries and the query are narrowed with
 * round-to-nearest-even (float16.go:53-63 BF16FromFloat32 incl. its NaN
 * rule; f32bitsToF16bits:114-160 == IEEE RTNE, which v_cvt_f16_f32
 * implements) and distances decode back to f32 and accumulate in f32
 * (distance_func_narrow.go:27-52). Layout: 16-byte units of u16 elements
 * ([g][dpad/8][64][8 halves]); dpad is padded to a multiple of 8 for
 * half-typed indexes. */

__device__ __forceinline__ float moann_h2f(uint16_t b) {
    __half_raw r;
    r.x = b;
    return __half2float((__half)r);
}
__device__ __forceinline__ float moann_bf2f(uint16_t b) {
    return __uint_as_float((uint32_t)b << 16);
}
__device__ __forceinline__ uint16_t moann_f2h(float f) {
    __half_raw r = (__half_raw)__float2half(f); /* RTNE */
    return r.x;
}
__device__ __forceinline__ uint16_t moann_f2bf(float f) {
    const uint32_t x = __float_as_uint(f);
    if (((x >> 23) & 0xff) == 0xff && (x & 0x7fffff))
        return (uint16_t)((x >> 16) | 0x0040); /* quiet the NaN */
    return (uint16_t)((x + 0x7fffu + ((x >> 16) & 1)) >> 16);
}

/* f32 rows -> u16 rows (pad columns 0); mirrors quantize_rows_kernel */
template <bool BF>
__global__ void quantize_half_rows_kernel(const float* __restrict__ in,
                                          int64_t nrows, int in_stride,
                                          int dim, int dpad,
                                          uint16_t* __restrict__ out) {
    const int64_t total = nrows * (int64_t)dpad;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x; /* see
        quantize_rows_kernel: 32-bit work-item grids */
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int64_t r = idx / dpad;
        const int c = (int)(idx % dpad);
        uint16_t v = 0;
        if (c < dim) {
            const float x = in[r * (int64_t)in_stride + c];
            v = BF ? moann_f2bf(x) : moann_f2h(x);
        }
        out[idx] = v;
    }
}

void launch_quantize_half_rows(bool bf, const float* in, int64_t nrows,
                               int in_stride, int dim, int dpad,
                               uint16_t* out, hipStream_t stream) {
    const int64_t total = nrows * (int64_t)dpad;
    if (!total) return;
    const int64_t grid = std::min<int64_t>((total + 255) / 256, 1 << 22);
    if (bf)
        hipLaunchKernelGGL(quantize_half_rows_kernel<true>,
                           dim3((uint32_t)grid), dim3(256), 0, stream, in,
                           nrows, in_stride, dim, dpad, out);
    else
        hipLaunchKernelGGL(quantize_half_rows_kernel<false>,
                           dim3((uint32_t)grid), dim3(256), 0, stream, in,
                           nrows, in_stride, dim, dpad, out);
}

/* |q|^2 over DECODED query halves (cosine denominators) */
template <bool BF>
__global__ void qnorms_h_kernel(const uint16_t* __restrict__ q, int nq,
                                int dpad, float* __restrict__ out) {
    const int i = blockIdx.x;
    const int lane = threadIdx.x & 63;
    if (threadIdx.x >= 64 || i >= nq) return;
    float acc = 0.f;
    for (int t = lane; t < dpad; t += 64) {
        const float x = BF ? moann_bf2f(q[(int64_t)i * dpad + t])
                           : moann_h2f(q[(int64_t)i * dpad + t]);
        acc = fmaf(x, x, acc);
    }
    for (int off = 32; off; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (lane == 0) out[i] = acc;
}

void launch_qnorms_h(bool bf, const uint16_t* q, int nq, int dpad, float* out,
                     hipStream_t stream) {
    if (!nq) return;
    if (bf)
        hipLaunchKernelGGL(qnorms_h_kernel<true>, dim3(nq), dim3(64), 0,
                           stream, q, nq, dpad, out);
    else
        hipLaunchKernelGGL(qnorms_h_kernel<false>, dim3(nq), dim3(64), 0,
                           stream, q, nq, dpad, out);
}

/* half-storage list scan: dwordx4 = 8 halves/lane per step, decoded to
 * f32, same job/filter/output semantics as the other scans. QT=8, GP=2. */
template <int METRIC, bool BF>
__global__ __launch_bounds__(256) void scan_h_kernel(
    const uint16_t* __restrict__ packed, const uint16_t* __restrict__ queries_h,
    const float* __restrict__ qnorms, int dpad,
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gstart, const int32_t* __restrict__ j_gcount,
    const int32_t* __restrict__ j_rows, const int32_t* __restrict__ j_nq,
    const int32_t* __restrict__ j_qbase,
    const int32_t* __restrict__ qslot_query,
    const int64_t* __restrict__ qslot_outbase,
    const int64_t* __restrict__ j_slot_base,
    const uint32_t* __restrict__ filter_bitset,
    float* __restrict__ dists_out) {
    constexpr int QT = 8;
    extern __shared__ float ldsf[];
    uint16_t* ldsq = (uint16_t*)ldsf;           /* [QT][dpad] halves */
    float* ldsn = (float*)(ldsq + QT * dpad);

    int j;
    {
        const int nwg = gridDim.x, bid = blockIdx.x;
        const int q8 = nwg >> 3, r8 = nwg & 7;
        const int xcd = bid & 7, idx = bid >> 3;
        j = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
        if (j >= nwg) j = bid;
    }
    const int nq = j_nq[j];
    const int qbase = j_qbase[j];
    for (int t = 0; t < QT; ++t) {
        if (t < nq) {
            const int q = qslot_query[qbase + t];
            const uint16_t* src = queries_h + (int64_t)q * dpad;
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = src[e];
            if (METRIC == KM_COS && threadIdx.x == 0) ldsn[t] = qnorms[q];
        } else {
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = 0;
            if (METRIC == KM_COS && threadIdx.x == 0) ldsn[t] = 0.f;
        }
    }
    __syncthreads();

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gstart = j_gstart[j], gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];
    const int d8 = dpad >> 3;

#define MOANN_H8(w, out0, out1)                                                  {                                                                                const float va = BF ? moann_bf2f((uint16_t)((w) & 0xFFFF))                                       : moann_h2f((uint16_t)((w) & 0xFFFF));                   const float vb = BF ? moann_bf2f((uint16_t)((w) >> 16))                                          : moann_h2f((uint16_t)((w) >> 16));                      out0 = va;                                                                   out1 = vb;                                                               }

    for (int gp = gstart + 2 * wave; gp < gstart + gcount; gp += 8) {
        const int g0 = gp;
        const bool has1 = (gp + 1) < (gstart + gcount);
        const int g1 = has1 ? gp + 1 : gp;
        const uint4* __restrict__ d0 =
            (const uint4*)(packed + (baseg + g0) * (int64_t)64 * dpad) + lane;
        const uint4* __restrict__ d1 =
            (const uint4*)(packed + (baseg + g1) * (int64_t)64 * dpad) + lane;

        float acc0[QT], acc1[QT];
#pragma unroll
        for (int t = 0; t < QT; ++t) acc0[t] = acc1[t] = 0.f;
        float rn0 = 0.f, rn1 = 0.f;

#pragma unroll 2
        for (int j8 = 0; j8 < d8; ++j8) {
            const uint4 c0 = d0[j8 * 64];
            const uint4 c1 = d1[j8 * 64];
            float x0[8], x1[8];
            MOANN_H8(c0.x, x0[0], x0[1]) MOANN_H8(c0.y, x0[2], x0[3])
            MOANN_H8(c0.z, x0[4], x0[5]) MOANN_H8(c0.w, x0[6], x0[7])
            MOANN_H8(c1.x, x1[0], x1[1]) MOANN_H8(c1.y, x1[2], x1[3])
            MOANN_H8(c1.z, x1[4], x1[5]) MOANN_H8(c1.w, x1[6], x1[7])
            if (METRIC == KM_COS) {
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    rn0 = fmaf(x0[u], x0[u], rn0);
                    rn1 = fmaf(x1[u], x1[u], rn1);
                }
            }
#pragma unroll
            for (int t = 0; t < QT; ++t) {
                const uint4 qw =
                    *(const uint4*)(ldsq + t * dpad + j8 * 8);
                float qv[8];
                MOANN_H8(qw.x, qv[0], qv[1]) MOANN_H8(qw.y, qv[2], qv[3])
                MOANN_H8(qw.z, qv[4], qv[5]) MOANN_H8(qw.w, qv[6], qv[7])
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    if (METRIC == KM_L2SQ) {
                        const float e0 = x0[u] - qv[u];
                        acc0[t] = fmaf(e0, e0, acc0[t]);
                        const float e1 = x1[u] - qv[u];
                        acc1[t] = fmaf(e1, e1, acc1[t]);
                    } else if (METRIC == KM_IP || METRIC == KM_COS) {
                        acc0[t] = fmaf(x0[u], qv[u], acc0[t]);
                        acc1[t] = fmaf(x1[u], qv[u], acc1[t]);
                    } else {
                        acc0[t] += fabsf(x0[u] - qv[u]);
                        acc1[t] += fabsf(x1[u] - qv[u]);
                    }
                }
            }
        }
#undef MOANN_H8

        const int row0 = g0 * 64 + lane;
        const int row1 = g1 * 64 + lane;
        bool pass0 = true, pass1 = true;
        if (filter_bitset) {
            const int64_t sb = j_slot_base[j];
            if (row0 < rows) {
                const int64_t gs = sb + row0;
                pass0 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
            if (has1 && row1 < rows) {
                const int64_t gs = sb + row1;
                pass1 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
        }
#pragma unroll
        for (int t = 0; t < QT; ++t) {
            if (t >= nq) break;
            const int64_t ob = qslot_outbase[qbase + t];
            if (row0 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = -acc0[t];
                else if (METRIC == KM_COS) {
                    const double denom =
                        sqrt((double)rn0) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc0[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else dist = acc0[t];
                dists_out[ob + row0] = pass0 ? dist : FLT_MAX;
            }
            if (has1 && row1 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = -acc1[t];
                else if (METRIC == KM_COS) {
                    const double denom =
                        sqrt((double)rn1) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc1[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else dist = acc1[t];
                dists_out[ob + row1] = pass1 ? dist : FLT_MAX;
            }
        }
    }
}

template <int METRIC>
static void launch_scan_h_m(bool bf, const uint16_t* packed,
                            const uint16_t* queries_h, const float* qnorms,
                            int dpad, const ScanJobs& jb, float* dists_out,
                            hipStream_t stream,
                            const uint32_t* filter_bitset) {
    const dim3 grid(jb.njobs), block(256);
    const size_t shmem = 8 * dpad * 2 + 8 * 4 + 16;
#define HFARGS packed, queries_h, qnorms, dpad, jb.databaseg, jb.gstart, \
    jb.gcount, jb.rows, jb.nq, jb.qbase, jb.qslot_query, jb.qslot_outbase, \
    jb.slot_base, filter_bitset, dists_out
    if (bf)
        hipLaunchKernelGGL((scan_h_kernel<METRIC, true>), grid, block, shmem,
                           stream, HFARGS);
    else
        hipLaunchKernelGGL((scan_h_kernel<METRIC, false>), grid, block,
                           shmem, stream, HFARGS);
#undef HFARGS
}

void launch_scan_h(int metric, bool bf, const uint16_t* packed,
                   const uint16_t* queries_h, const float* qnorms, int dpad,
                   const ScanJobs& jb, float* dists_out, hipStream_t stream,
                   const uint32_t* filter_bitset) {
    if (!jb.njobs) return;
    switch (metric) {
    case KM_L2SQ: launch_scan_h_m<KM_L2SQ>(bf, packed, queries_h, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    case KM_IP:   launch_scan_h_m<KM_IP>(bf, packed, queries_h, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    case KM_COS:  launch_scan_h_m<KM_COS>(bf, packed, queries_h, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    default:      launch_scan_h_m<KM_L1>(bf, packed, queries_h, qnorms, dpad, jb, dists_out, stream, filter_bitset); break;
    }
}

/* ---------------- exact f32 refine (two-stage scan) ---------------------
 * Re-rank the byte-stage top-R candidates with exact f32 distances over a
 * row-major copy of the entries (SQ8-with-refine, the standard recall
 * recovery; the byte stage reads 1/4 the bytes of the f32 scan). One
 * workgroup per query: query row cached in LDS, one WAVE per candidate,
 * lane-parallel over dims, shfl reduce. Distance semantics identical to
 * the f32 scan kernel (MO conventions incl. cosine's double denom). */
template <int METRIC>
__global__ __launch_bounds__(256) void refine_kernel(
    const float* __restrict__ rows_f32, /* [count][dim] slot-major */
    const float* __restrict__ queries,  /* [nq][dpad] */
    const float* __restrict__ qnorms,   /* [nq] |q|^2 (cos) */
    int dim, int dpad, int R,
    const int32_t* __restrict__ rsel_slots, /* [nq][R] candidate positions */
    const float* __restrict__ rsel_dists,   /* [nq][R] byte dists (padding) */
    const int32_t* __restrict__ probe_lists,
    const int64_t* __restrict__ probe_offs,
    const int64_t* __restrict__ list_slot_base, int nprobe,
    float* __restrict__ refined /* [nq][R] */) {
    extern __shared__ float ldsq[]; /* [dpad] */
    const int q = blockIdx.x;
    const float* qv = queries + (int64_t)q * dpad;
    for (int e = threadIdx.x; e < dpad; e += blockDim.x) ldsq[e] = qv[e];
    __syncthreads();
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int64_t* offs = probe_offs + (int64_t)q * (nprobe + 1);

    for (int c = wave; c < R; c += 4) {
        const int64_t idx = (int64_t)q * R + c;
        const int32_t pos = rsel_slots[idx];
        if (pos < 0 || rsel_dists[idx] == FLT_MAX) {
            if (lane == 0) refined[idx] = FLT_MAX;
            continue;
        }
        /* candidate position -> (probed list, local row) -> global slot */
        int lo = 0, hi = nprobe - 1;
        while (lo < hi) {
            const int mid = (lo + hi + 1) >> 1;
            if (offs[mid] <= pos) lo = mid; else hi = mid - 1;
        }
        const int32_t list = probe_lists[(int64_t)q * nprobe + lo];
        const int64_t slot = list_slot_base[list] + (pos - offs[lo]);
        const float* row = rows_f32 + slot * (int64_t)dim;
        float acc = 0.f, rn = 0.f;
        for (int t = lane; t < dim; t += 64) {
            const float x = row[t], qq = ldsq[t];
            if (METRIC == KM_L2SQ) {
                const float e = x - qq;
                acc = fmaf(e, e, acc);
            } else if (METRIC == KM_IP || METRIC == KM_COS) {
                acc = fmaf(x, qq, acc);
                if (METRIC == KM_COS) rn = fmaf(x, x, rn);
            } else {
                acc += fabsf(x - qq);
            }
        }
        for (int off = 32; off; off >>= 1) {
            acc += __shfl_down(acc, off, 64);
            if (METRIC == KM_COS) rn += __shfl_down(rn, off, 64);
        }
        if (lane == 0) {
            float dist;
            if (METRIC == KM_IP) dist = -acc;
            else if (METRIC == KM_COS) {
                const double denom = sqrt((double)rn) * sqrt((double)qnorms[q]);
                if (denom == 0.0) dist = 1.0f;
                else {
                    double sim = (double)acc / denom;
                    sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                    dist = (float)(1.0 - sim);
                }
            } else dist = acc;
            refined[idx] = dist;
        }
    }
}

void launch_refine(int metric, const float* rows_f32, const float* queries,
                   const float* qnorms, int dim, int dpad, int R, int nq,
                   const int32_t* rsel_slots, const float* rsel_dists,
                   const int32_t* probe_lists, const int64_t* probe_offs,
                   const int64_t* list_slot_base, int nprobe, float* refined,
                   hipStream_t stream) {
    if (!nq) return;
    const size_t shmem = (size_t)dpad * 4;
#define RARGS rows_f32, queries, qnorms, dim, dpad, R, rsel_slots, \
    rsel_dists, probe_lists, probe_offs, list_slot_base, nprobe, refined
    switch (metric) {
    case KM_L2SQ: hipLaunchKernelGGL(refine_kernel<KM_L2SQ>, dim3(nq), dim3(256), shmem, stream, RARGS); break;
    case KM_IP:   hipLaunchKernelGGL(refine_kernel<KM_IP>, dim3(nq), dim3(256), shmem, stream, RARGS); break;
    case KM_COS:  hipLaunchKernelGGL(refine_kernel<KM_COS>, dim3(nq), dim3(256), shmem, stream, RARGS); break;
    default:      hipLaunchKernelGGL(refine_kernel<KM_L1>, dim3(nq), dim3(256), shmem, stream, RARGS); break;
    }
#undef RARGS
}

/* final-selection composition: sel2 indexes into the refine set; map back
 * to candidate positions so the shared gather kernel applies unchanged */
__global__ void compose_select_kernel(const int32_t* __restrict__ sel2,
                                      const int32_t* __restrict__ rsel_slots,
                                      int R, int k, int64_t total,
                                      int32_t* __restrict__ out) {
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= total) return;
    const int32_t s2 = sel2[i];
    const int64_t q = i / k;
    out[i] = s2 < 0 ? -1 : rsel_slots[q * R + s2];
}

void launch_compose_select(const int32_t* sel2, const int32_t* rsel_slots,
                           int R, int k, int nq, int32_t* out,
                           hipStream_t stream) {
    const int64_t total = (int64_t)nq * k;
    if (!total) return;
    hipLaunchKernelGGL(compose_select_kernel,
                       dim3((uint32_t)((total + 255) / 256)), dim3(256), 0,
                       stream, sel2, rsel_slots, R, k, total, out);
}

/* -------------------------- quantized (int8/uint8) scan ------------------
 * Narrow-storage variant of the list scan (reference: int8/uint8 entries
 * quantized by q(x)=round(x*mul+add), distance computed IN THE QUANTIZED
 * DOMAIN with exact integer accumulation — distance_func_narrow.go:330-450;
 * the /mul^2 rescale happens at gather, ivfflat/search.go:1062-1077).
 * Data layout: bytes [g][dpad/4][64][4] (same interleave as f32, byte
 * elements — one coalesced 256 B read per wave step). Query tile: quantized
 * bytes in LDS. int32 accumulators are exact for dpad <= 33024 (l2sq term
 * <= 65025/elem). */
template <int METRIC, bool UNSIGNED>
__global__ __launch_bounds__(256) void scan_i8_kernel(
    const uint8_t* __restrict__ packed, const uint8_t* __restrict__ queries_q,
    const int32_t* __restrict__ qnorms_i /* [nq] sum sq, cos only */,
    int dpad,
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gstart, const int32_t* __restrict__ j_gcount,
    const int32_t* __restrict__ j_rows, const int32_t* __restrict__ j_nq,
    const int32_t* __restrict__ j_qbase,
    const int32_t* __restrict__ qslot_query,
    const int64_t* __restrict__ qslot_outbase,
    const int64_t* __restrict__ j_slot_base,
    const uint32_t* __restrict__ filter_bitset,
    float* __restrict__ dists_out) {
    constexpr int QT = 8;
    extern __shared__ float ldsf[];
    uint8_t* ldsq = (uint8_t*)ldsf;            /* [QT][dpad] bytes */
    int32_t* ldsn = (int32_t*)(ldsq + QT * dpad);

    int j;
    {
        const int nwg = gridDim.x, bid = blockIdx.x;
        const int q8 = nwg >> 3, r8 = nwg & 7;
        const int xcd = bid & 7, idx = bid >> 3;
        j = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
        if (j >= nwg) j = bid;
    }
    const int nq = j_nq[j];
    const int qbase = j_qbase[j];
    for (int t = 0; t < QT; ++t) {
        if (t < nq) {
            const int q = qslot_query[qbase + t];
            const uint8_t* src = queries_q + (int64_t)q * dpad;
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = src[e];
            if (METRIC == KM_COS && threadIdx.x == 0) ldsn[t] = qnorms_i[q];
        } else {
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = 0;
            if (METRIC == KM_COS && threadIdx.x == 0) ldsn[t] = 0;
        }
    }
    __syncthreads();

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gstart = j_gstart[j], gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];
    const int d4 = dpad >> 2;

    const auto sx = [](uint8_t b) -> int {
        return UNSIGNED ? (int)b : (int)(int8_t)b;
    };

    for (int gp = gstart + 2 * wave; gp < gstart + gcount; gp += 8) {
        const int g0 = gp;
        const bool has1 = (gp + 1) < (gstart + gcount);
        const int g1 = has1 ? gp + 1 : gp;
        const uchar4* __restrict__ d0 =
            (const uchar4*)(packed + (baseg + g0) * (int64_t)64 * dpad) + lane;
        const uchar4* __restrict__ d1 =
            (const uchar4*)(packed + (baseg + g1) * (int64_t)64 * dpad) + lane;
        int32_t acc0[QT], acc1[QT];
#pragma unroll
        for (int t = 0; t < QT; ++t) acc0[t] = acc1[t] = 0;
        int32_t rn0 = 0, rn1 = 0;
#pragma unroll 2
        for (int q4 = 0; q4 < d4; ++q4) {
            const uchar4 x0 = d0[q4 * 64];
            const uchar4 x1 = d1[q4 * 64];
            if (METRIC == KM_COS) {
                rn0 += sx(x0.x) * sx(x0.x) + sx(x0.y) * sx(x0.y) +
                       sx(x0.z) * sx(x0.z) + sx(x0.w) * sx(x0.w);
                rn1 += sx(x1.x) * sx(x1.x) + sx(x1.y) * sx(x1.y) +
                       sx(x1.z) * sx(x1.z) + sx(x1.w) * sx(x1.w);
            }
#pragma unroll
            for (int t = 0; t < QT; ++t) {
                const uchar4 qv = ((const uchar4*)(ldsq + t * dpad))[q4];
                if (METRIC == KM_L2SQ) {
                    int e;
                    e = sx(x0.x) - sx(qv.x); acc0[t] += e * e;
                    e = sx(x0.y) - sx(qv.y); acc0[t] += e * e;
                    e = sx(x0.z) - sx(qv.z); acc0[t] += e * e;
                    e = sx(x0.w) - sx(qv.w); acc0[t] += e * e;
                    e = sx(x1.x) - sx(qv.x); acc1[t] += e * e;
                    e = sx(x1.y) - sx(qv.y); acc1[t] += e * e;
                    e = sx(x1.z) - sx(qv.z); acc1[t] += e * e;
                    e = sx(x1.w) - sx(qv.w); acc1[t] += e * e;
                } else if (METRIC == KM_IP || METRIC == KM_COS) {
                    acc0[t] += sx(x0.x) * sx(qv.x) + sx(x0.y) * sx(qv.y) +
                               sx(x0.z) * sx(qv.z) + sx(x0.w) * sx(qv.w);
                    acc1[t] += sx(x1.x) * sx(qv.x) + sx(x1.y) * sx(qv.y) +
                               sx(x1.z) * sx(qv.z) + sx(x1.w) * sx(qv.w);
                } else { /* KM_L1 */
                    acc0[t] += abs(sx(x0.x) - sx(qv.x)) +
                               abs(sx(x0.y) - sx(qv.y)) +
                               abs(sx(x0.z) - sx(qv.z)) +
                               abs(sx(x0.w) - sx(qv.w));
                    acc1[t] += abs(sx(x1.x) - sx(qv.x)) +
                               abs(sx(x1.y) - sx(qv.y)) +
                               abs(sx(x1.z) - sx(qv.z)) +
                               abs(sx(x1.w) - sx(qv.w));
                }
            }
        }
        const int row0 = g0 * 64 + lane;
        const int row1 = g1 * 64 + lane;
        bool pass0 = true, pass1 = true;
        if (filter_bitset) {
            const int64_t sb = j_slot_base[j];
            if (row0 < rows) {
                const int64_t gs = sb + row0;
                pass0 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
            if (has1 && row1 < rows) {
                const int64_t gs = sb + row1;
                pass1 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
        }
#pragma unroll
        for (int t = 0; t < QT; ++t) {
            if (t >= nq) break;
            const int64_t ob = qslot_outbase[qbase + t];
            if (row0 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = (float)(-(int64_t)acc0[t]);
                else if (METRIC == KM_COS) {
                    /* distance_func_narrow.go:444-449: int sums -> double */
                    const double denom =
                        sqrt((double)rn0) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc0[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else dist = (float)acc0[t];
                dists_out[ob + row0] = pass0 ? dist : FLT_MAX;
            }
            if (has1 && row1 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = (float)(-(int64_t)acc1[t]);
                else if (METRIC == KM_COS) {
                    const double denom =
                        sqrt((double)rn1) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc1[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else dist = (float)acc1[t];
                dists_out[ob + row1] = pass1 ? dist : FLT_MAX;
            }
        }
    }
}

/* Dot-form byte scan (dpad % 16 == 0): one dwordx4 load = 16 dims of the
 * row per lane (1 KiB per wave step, same stream shape as the f32 kernel
 * — the uchar4 kernel's 256 B loads were MLP-starved at ~1 TB/s), and the
 * arithmetic uses the CDNA dot/sad byte ops:
 *   L2sq:  |x-q|^2 = rn + qn - 2*dot(x,q)  (exact in int32: bounds
 *          255^2*dpad <= 5.1e7 at dpad<=768; v_dot4_{i32_i8,u32_u8})
 *   IP/cos: dot(x,q) directly (cos norms precomputed, same ints)
 *   L1:    v_sad_u8 (signed values biased by XOR 0x80 per byte — two's
 *          complement int8 + 128 == x ^ 0x80, |x-q| unchanged)
 * Row norms rn come from launch_rownorms_i8 at build time; query norms qn
 * from launch_qnorms_i8. Results are bit-identical to scan_i8_kernel
 * (exact integer arithmetic both ways); that kernel stays the path for
 * dpad % 16 != 0. */
template <int METRIC, bool UNSIGNED, int QT = 8>
__global__ __launch_bounds__(256) void scan_i8_dot_kernel(
    const uint8_t* __restrict__ packed, const uint8_t* __restrict__ queries_q,
    const int32_t* __restrict__ qnorms_i, /* [nq] sum sq (l2/cos) */
    const int32_t* __restrict__ rownorms, /* [ngroups*64] sum sq (l2/cos) */
    int dpad,
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gstart, const int32_t* __restrict__ j_gcount,
    const int32_t* __restrict__ j_rows, const int32_t* __restrict__ j_nq,
    const int32_t* __restrict__ j_qbase,
    const int32_t* __restrict__ qslot_query,
    const int64_t* __restrict__ qslot_outbase,
    const int64_t* __restrict__ j_slot_base,
    const uint32_t* __restrict__ filter_bitset,
    float* __restrict__ dists_out) {
    constexpr bool NEED_N = METRIC == KM_L2SQ || METRIC == KM_COS;
    extern __shared__ float ldsf[];
    uint8_t* ldsq = (uint8_t*)ldsf;            /* [QT][dpad] bytes */
    int32_t* ldsn = (int32_t*)(ldsq + QT * dpad);

    int j;
    {
        const int nwg = gridDim.x, bid = blockIdx.x;
        const int q8 = nwg >> 3, r8 = nwg & 7;
        const int xcd = bid & 7, idx = bid >> 3;
        j = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
        if (j >= nwg) j = bid;
    }
    const int nq = j_nq[j];
    const int qbase = j_qbase[j];
    for (int t = 0; t < QT; ++t) {
        if (t < nq) {
            const int q = qslot_query[qbase + t];
            const uint8_t* src = queries_q + (int64_t)q * dpad;
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = src[e];
            if (NEED_N && threadIdx.x == 0) ldsn[t] = qnorms_i[q];
        } else {
            for (int e = threadIdx.x; e < dpad; e += blockDim.x)
                ldsq[t * dpad + e] = 0;
            if (NEED_N && threadIdx.x == 0) ldsn[t] = 0;
        }
    }
    __syncthreads();

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gstart = j_gstart[j], gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];
    const int d16 = dpad >> 4;

    for (int gp = gstart + 2 * wave; gp < gstart + gcount; gp += 8) {
        const int g0 = gp;
        const bool has1 = (gp + 1) < (gstart + gcount);
        const int g1 = has1 ? gp + 1 : gp;
        const uint4* __restrict__ d0 =
            (const uint4*)(packed + (baseg + g0) * (int64_t)64 * dpad) + lane;
        const uint4* __restrict__ d1 =
            (const uint4*)(packed + (baseg + g1) * (int64_t)64 * dpad) + lane;

        int32_t acc0[QT], acc1[QT];
#pragma unroll
        for (int t = 0; t < QT; ++t) acc0[t] = acc1[t] = 0;

#pragma unroll 2
        for (int j16 = 0; j16 < d16; ++j16) {
            uint4 x0 = d0[j16 * 64];
            uint4 x1 = d1[j16 * 64];
            if (METRIC == KM_L1 && !UNSIGNED) {
                /* bias int8 -> u8 so v_sad_u8 applies */
                x0.x ^= 0x80808080u; x0.y ^= 0x80808080u;
                x0.z ^= 0x80808080u; x0.w ^= 0x80808080u;
                x1.x ^= 0x80808080u; x1.y ^= 0x80808080u;
                x1.z ^= 0x80808080u; x1.w ^= 0x80808080u;
            }
#pragma unroll
            for (int t = 0; t < QT; ++t) {
                uint4 qv = *(const uint4*)(ldsq + t * dpad + j16 * 16);
                if (METRIC == KM_L1) {
                    if (!UNSIGNED) {
                        qv.x ^= 0x80808080u; qv.y ^= 0x80808080u;
                        qv.z ^= 0x80808080u; qv.w ^= 0x80808080u;
                    }
                    acc0[t] = (int32_t)__builtin_amdgcn_sad_u8(x0.x, qv.x, (uint32_t)acc0[t]);
                    acc0[t] = (int32_t)__builtin_amdgcn_sad_u8(x0.y, qv.y, (uint32_t)acc0[t]);
                    acc0[t] = (int32_t)__builtin_amdgcn_sad_u8(x0.z, qv.z, (uint32_t)acc0[t]);
                    acc0[t] = (int32_t)__builtin_amdgcn_sad_u8(x0.w, qv.w, (uint32_t)acc0[t]);
                    acc1[t] = (int32_t)__builtin_amdgcn_sad_u8(x1.x, qv.x, (uint32_t)acc1[t]);
                    acc1[t] = (int32_t)__builtin_amdgcn_sad_u8(x1.y, qv.y, (uint32_t)acc1[t]);
                    acc1[t] = (int32_t)__builtin_amdgcn_sad_u8(x1.z, qv.z, (uint32_t)acc1[t]);
                    acc1[t] = (int32_t)__builtin_amdgcn_sad_u8(x1.w, qv.w, (uint32_t)acc1[t]);
                } else if (UNSIGNED) {
                    acc0[t] = (int32_t)__builtin_amdgcn_udot4(x0.x, qv.x, (uint32_t)acc0[t], false);
                    acc0[t] = (int32_t)__builtin_amdgcn_udot4(x0.y, qv.y, (uint32_t)acc0[t], false);
                    acc0[t] = (int32_t)__builtin_amdgcn_udot4(x0.z, qv.z, (uint32_t)acc0[t], false);
                    acc0[t] = (int32_t)__builtin_amdgcn_udot4(x0.w, qv.w, (uint32_t)acc0[t], false);
                    acc1[t] = (int32_t)__builtin_amdgcn_udot4(x1.x, qv.x, (uint32_t)acc1[t], false);
                    acc1[t] = (int32_t)__builtin_amdgcn_udot4(x1.y, qv.y, (uint32_t)acc1[t], false);
                    acc1[t] = (int32_t)__builtin_amdgcn_udot4(x1.z, qv.z, (uint32_t)acc1[t], false);
                    acc1[t] = (int32_t)__builtin_amdgcn_udot4(x1.w, qv.w, (uint32_t)acc1[t], false);
                } else {
                    acc0[t] = __builtin_amdgcn_sdot4((int32_t)x0.x, (int32_t)qv.x, acc0[t], false);
                    acc0[t] = __builtin_amdgcn_sdot4((int32_t)x0.y, (int32_t)qv.y, acc0[t], false);
                    acc0[t] = __builtin_amdgcn_sdot4((int32_t)x0.z, (int32_t)qv.z, acc0[t], false);
                    acc0[t] = __builtin_amdgcn_sdot4((int32_t)x0.w, (int32_t)qv.w, acc0[t], false);
                    acc1[t] = __builtin_amdgcn_sdot4((int32_t)x1.x, (int32_t)qv.x, acc1[t], false);
                    acc1[t] = __builtin_amdgcn_sdot4((int32_t)x1.y, (int32_t)qv.y, acc1[t], false);
                    acc1[t] = __builtin_amdgcn_sdot4((int32_t)x1.z, (int32_t)qv.z, acc1[t], false);
                    acc1[t] = __builtin_amdgcn_sdot4((int32_t)x1.w, (int32_t)qv.w, acc1[t], false);
                }
            }
        }

        const int row0 = g0 * 64 + lane;
        const int row1 = g1 * 64 + lane;
        const int32_t rn0 = NEED_N ? rownorms[(baseg + g0) * 64 + lane] : 0;
        const int32_t rn1 = NEED_N ? rownorms[(baseg + g1) * 64 + lane] : 0;
        bool pass0 = true, pass1 = true;
        if (filter_bitset) {
            const int64_t sb = j_slot_base[j];
            if (row0 < rows) {
                const int64_t gs = sb + row0;
                pass0 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
            if (has1 && row1 < rows) {
                const int64_t gs = sb + row1;
                pass1 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
            }
        }
#pragma unroll
        for (int t = 0; t < QT; ++t) {
            if (t >= nq) break;
            const int64_t ob = qslot_outbase[qbase + t];
            if (row0 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = (float)(-(int64_t)acc0[t]);
                else if (METRIC == KM_COS) {
                    const double denom =
                        sqrt((double)rn0) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc0[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else if (METRIC == KM_L2SQ)
                    dist = (float)(rn0 + ldsn[t] - 2 * acc0[t]);
                else dist = (float)acc0[t]; /* L1 */
                dists_out[ob + row0] = pass0 ? dist : FLT_MAX;
            }
            if (has1 && row1 < rows) {
                float dist;
                if (METRIC == KM_IP) dist = (float)(-(int64_t)acc1[t]);
                else if (METRIC == KM_COS) {
                    const double denom =
                        sqrt((double)rn1) * sqrt((double)ldsn[t]);
                    if (denom == 0.0) dist = 1.0f;
                    else {
                        double sim = (double)acc1[t] / denom;
                        sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                        dist = (float)(1.0 - sim);
                    }
                } else if (METRIC == KM_L2SQ)
                    dist = (float)(rn1 + ldsn[t] - 2 * acc1[t]);
                else dist = (float)acc1[t];
                dists_out[ob + row1] = pass1 ? dist : FLT_MAX;
            }
        }
    }
}

/* pack quantized byte rows into 16-byte units: [g][dpad/16][64][16] — the
 * layout the dot-form scan streams with 16 B/lane dwordx4 loads (the
 * uchar4 [g][dpad/4][64][4] layout only gives 4 B/lane per 256 B block).
 * Used when dpad % 16 == 0; thread per output uint4. */
__global__ void bytes_pack16_kernel(const uint8_t* __restrict__ rows_q,
                                    int dpad,
                                    const int64_t* __restrict__ group_rowbase,
                                    const int32_t* __restrict__ group_valid,
                                    const int64_t* __restrict__ slot_rows,
                                    int64_t ngroups,
                                    uint8_t* __restrict__ packed) {
    const int d16 = dpad >> 4;
    const int64_t nunits = ngroups * (int64_t)d16 * 64;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < nunits; idx += stride) {
        const int lane = (int)(idx & 63);
        const int64_t rest = idx >> 6;
        const int m16 = (int)(rest % d16);
        const int64_t g = rest / d16;
        uint4 v = {0, 0, 0, 0};
        if (lane < group_valid[g]) {
            const int64_t row = slot_rows[group_rowbase[g] + lane];
            v = *(const uint4*)(rows_q + row * (int64_t)dpad + 16 * m16);
        }
        ((uint4*)packed)[idx] = v;
    }
}

void launch_bytes_pack16(const uint8_t* rows_q, int dpad,
                         const int64_t* group_rowbase,
                         const int32_t* group_valid,
                         const int64_t* slot_rows, int64_t ngroups,
                         uint8_t* packed, hipStream_t stream) {
    const int64_t nunits = ngroups * (int64_t)(dpad >> 4) * 64;
    if (!nunits) return;
    const int block = 256;
    const int64_t grid = std::min<int64_t>((nunits + block - 1) / block,
                                           1 << 22);
    hipLaunchKernelGGL(bytes_pack16_kernel, dim3((uint32_t)grid), dim3(block),
                       0, stream, rows_q, dpad, group_rowbase, group_valid,
                       slot_rows, ngroups, packed);
}

/* per-(group,lane) byte-row sum of squares, same wave layout as the scan */
template <bool UNSIGNED>
__global__ __launch_bounds__(256) void rownorms_i8_kernel(
    const uint8_t* __restrict__ packed, int64_t ngroups, int dpad,
    int32_t* __restrict__ out) {
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int64_t g = (int64_t)blockIdx.x * 4 + wave;
    if (g >= ngroups) return;
    /* layout16: [g][dpad/16][64][16] (bytes_pack16_kernel) */
    const uint4* __restrict__ d =
        (const uint4*)(packed + g * (int64_t)64 * dpad) + lane;
    int32_t acc = 0;
    const int d16 = dpad >> 4;
    for (int m = 0; m < d16; ++m) {
        const uint4 v = d[m * 64];
        if (UNSIGNED) {
            acc = (int32_t)__builtin_amdgcn_udot4(v.x, v.x, (uint32_t)acc, false);
            acc = (int32_t)__builtin_amdgcn_udot4(v.y, v.y, (uint32_t)acc, false);
            acc = (int32_t)__builtin_amdgcn_udot4(v.z, v.z, (uint32_t)acc, false);
            acc = (int32_t)__builtin_amdgcn_udot4(v.w, v.w, (uint32_t)acc, false);
        } else {
            acc = __builtin_amdgcn_sdot4((int32_t)v.x, (int32_t)v.x, acc, false);
            acc = __builtin_amdgcn_sdot4((int32_t)v.y, (int32_t)v.y, acc, false);
            acc = __builtin_amdgcn_sdot4((int32_t)v.z, (int32_t)v.z, acc, false);
            acc = __builtin_amdgcn_sdot4((int32_t)v.w, (int32_t)v.w, acc, false);
        }
    }
    out[g * 64 + lane] = acc;
}

void launch_rownorms_i8(bool uns, const uint8_t* packed, int64_t ngroups,
                        int dpad, int32_t* out, hipStream_t stream) {
    const dim3 grid((uint32_t)((ngroups + 3) / 4)), block(256);
    if (uns)
        hipLaunchKernelGGL((rownorms_i8_kernel<true>), grid, block, 0, stream,
                           packed, ngroups, dpad, out);
    else
        hipLaunchKernelGGL((rownorms_i8_kernel<false>), grid, block, 0, stream,
                           packed, ngroups, dpad, out);
}

template <int METRIC>
static void launch_scan_i8_m(bool uns, const uint8_t* packed,
                             const uint8_t* queries_q, const int32_t* qnorms,
                             const int32_t* rownorms, int dpad,
                             const ScanJobs& jb, float* dists_out,
                             hipStream_t stream,
                             const uint32_t* filter_bitset, int qt) {
    const dim3 grid(jb.njobs), block(256);
    const size_t shmem = (size_t)qt * dpad + (size_t)qt * 4 + 16;
    /* dot-form kernel needs 16-dim-aligned rows and the precomputed row
     * norms; falls back to the uchar4 kernel otherwise (odd dims).
     * qt MUST match the job builder's tile width (the r01 refine bug). */
    const bool dot = (dpad & 15) == 0 && rownorms != nullptr;
#define I8ARGS packed, queries_q, qnorms, dpad, jb.databaseg, jb.gstart, \
    jb.gcount, jb.rows, jb.nq, jb.qbase, jb.qslot_query, jb.qslot_outbase, \
    jb.slot_base, filter_bitset, dists_out
#define I8DARGS packed, queries_q, qnorms, rownorms, dpad, jb.databaseg, \
    jb.gstart, jb.gcount, jb.rows, jb.nq, jb.qbase, jb.qslot_query, \
    jb.qslot_outbase, jb.slot_base, filter_bitset, dists_out
    if (dot && qt == 16) {
        if (uns)
            hipLaunchKernelGGL((scan_i8_dot_kernel<METRIC, true, 16>), grid,
                               block, shmem, stream, I8DARGS);
        else
            hipLaunchKernelGGL((scan_i8_dot_kernel<METRIC, false, 16>), grid,
                               block, shmem, stream, I8DARGS);
    } else if (dot && qt == 8) {
        if (uns)
            hipLaunchKernelGGL((scan_i8_dot_kernel<METRIC, true, 8>), grid,
                               block, shmem, stream, I8DARGS);
        else
            hipLaunchKernelGGL((scan_i8_dot_kernel<METRIC, false, 8>), grid,
                               block, shmem, stream, I8DARGS);
    } else if (qt != 8) {
        /* no QT!=8 instantiation of the uchar4 kernel: loud geometry error
         * instead of silently unscanned tile lanes */
        (void)hipErrorInvalidValue;
        hipLaunchKernelGGL((scan_i8_kernel<METRIC, true>), dim3(0), block,
                           shmem, stream, I8ARGS); /* grid 0 -> launch err */
    } else if (uns)
        hipLaunchKernelGGL((scan_i8_kernel<METRIC, true>), grid, block, shmem,
                           stream, I8ARGS);
    else
        hipLaunchKernelGGL((scan_i8_kernel<METRIC, false>), grid, block,
                           shmem, stream, I8ARGS);
#undef I8ARGS
#undef I8DARGS
}

void launch_scan_i8(int metric, bool uns, const uint8_t* packed,
                    const uint8_t* queries_q, const int32_t* qnorms,
                    const int32_t* rownorms, int dpad,
                    const ScanJobs& jb, float* dists_out, hipStream_t stream,
                    const uint32_t* filter_bitset, int qt) {
    if (!jb.njobs) return;
    switch (metric) {
    case KM_L2SQ: launch_scan_i8_m<KM_L2SQ>(uns, packed, queries_q, qnorms, rownorms, dpad, jb, dists_out, stream, filter_bitset, qt); break;
    case KM_IP:   launch_scan_i8_m<KM_IP>(uns, packed, queries_q, qnorms, rownorms, dpad, jb, dists_out, stream, filter_bitset, qt); break;
    case KM_COS:  launch_scan_i8_m<KM_COS>(uns, packed, queries_q, qnorms, rownorms, dpad, jb, dists_out, stream, filter_bitset, qt); break;
    default:      launch_scan_i8_m<KM_L1>(uns, packed, queries_q, qnorms, rownorms, dpad, jb, dists_out, stream, filter_bitset, qt); break;
    }
}

/* quantize f32 rows -> int8/uint8 with the reference's exact semantics:
 * two separate f32 roundings (product, then sum — quantizer.go:165-176),
 * round half away from zero, clamp, NaN -> 0 (float16.go:227-239).
 * Pad columns (dim <= c < dpad) emit 0 so padded dims stay identity for
 * every metric (q(0) is generally nonzero — quantizing pads would poison
 * IP/cosine). in: [rows][in_stride] f32; out: [rows][dpad] bytes. */
template <bool UNSIGNED>
__global__ void quantize_rows_kernel(const float* __restrict__ in,
                                     int64_t nrows, int in_stride, int dim,
                                     int dpad, float fmul, float fadd,
                                     uint8_t* __restrict__ out) {
    /* GRID-STRIDE: the AQL packet's grid size counts WORK-ITEMS in 32
     * bits, so a flat launch of >= 2^32 threads silently wraps (measured:
     * exactly total mod 2^32 elements written at 6M x 768 halves). Every
     * elementwise kernel over rows x dpad strides instead. */
    const int64_t total = nrows * (int64_t)dpad;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < total; i += stride) {
        const int64_t r = i / dpad;
        const int c = (int)(i % dpad);
        if (c >= dim) {
            out[i] = 0;
            continue;
        }
        const float x = in[r * (int64_t)in_stride + c];
        float m = x * fmul;
        asm volatile("" : "+v"(m)); /* block FMA fusion of the two f32 ops */
        const float sum = m + fadd;
        float v;
        if (sum != sum) v = 0.f;
        else v = (float)round((double)sum);
        const float lo = UNSIGNED ? 0.f : -128.f;
        const float hi = UNSIGNED ? 255.f : 127.f;
        v = v < lo ? lo : (v > hi ? hi : v);
        out[i] = UNSIGNED ? (uint8_t)v : (uint8_t)(int8_t)v;
    }
}

void launch_quantize_rows(bool uns, const float* in, int64_t nrows,
                          int in_stride, int dim, int dpad, float fmul,
                          float fadd, uint8_t* out, hipStream_t stream) {
    const int64_t total = nrows * (int64_t)dpad;
    if (!total) return;
    const int block = 256;
    /* cap work-items below 2^31: the AQL grid is 32-bit in WORK-ITEMS */
    const int64_t grid = std::min<int64_t>((total + block - 1) / block,
                                           1 << 22);
    if (uns)
        hipLaunchKernelGGL((quantize_rows_kernel<true>), dim3((uint32_t)grid),
                           dim3(block), 0, stream, in, nrows, in_stride, dim,
                           dpad, fmul, fadd, out);
    else
        hipLaunchKernelGGL((quantize_rows_kernel<false>), dim3((uint32_t)grid),
                           dim3(block), 0, stream, in, nrows, in_stride, dim,
                           dpad, fmul, fadd, out);
}

/* int32 sum-of-squares per quantized query (cos, narrow path) */
template <bool UNSIGNED>
__global__ void qnorm_i8_kernel(const uint8_t* __restrict__ q, int nq,
                                int dpad, int32_t* __restrict__ out) {
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= nq) return;
    const uint8_t* row = q + (int64_t)i * dpad;
    int32_t s = 0;
    for (int e = 0; e < dpad; ++e) {
        const int v = UNSIGNED ? (int)row[e] : (int)(int8_t)row[e];
        s += v * v;
    }
    out[i] = s;
}

void launch_qnorms_i8(bool uns, const uint8_t* q, int nq, int dpad,
                      int32_t* out, hipStream_t stream) {
    if (!nq) return;
    const int grid = (nq + 255) / 256;
    if (uns)
        hipLaunchKernelGGL((qnorm_i8_kernel<true>), dim3(grid), dim3(256), 0,
                           stream, q, nq, dpad, out);
    else
        hipLaunchKernelGGL((qnorm_i8_kernel<false>), dim3(grid), dim3(256), 0,
                           stream, q, nq, dpad, out);
}

/* --------------------------- pairwise 1xN --------------------------------
 * The SQL distance-builtin batch (metric.PairwiseDistanceLaunch,
 * pkg/sql/plan/function/func_binary.go:9127 const-query 1xN) and the legacy
 * cgo/cuda/mocl.cu l2distance kernels, subsumed: one wave per row over
 * row-major data, MO distance conventions. */
template <int METRIC>
__global__ __launch_bounds__(256) void pairwise_kernel(
    const float* __restrict__ rows, const float* __restrict__ query,
    float qnorm /* |q|^2, cos only */, int64_t n, int dim,
    float* __restrict__ out) {
    const int64_t r = (int64_t)blockIdx.x * (blockDim.x >> 6) +
                      (threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    if (r >= n) return;
    const float* row = rows + r * (int64_t)dim;
    float acc = 0.f, nrm = 0.f;
    for (int e = lane; e < dim; e += 64) {
        const float x = row[e], qv = query[e];
        if (METRIC == KM_L2SQ) {
            const float d = x - qv;
            acc = fmaf(d, d, acc);
        } else if (METRIC == KM_L1) {
            acc += fabsf(x - qv);
        } else {
            acc = fmaf(x, qv, acc);
            if (METRIC == KM_COS) nrm += x * x;
        }
    }
#pragma unroll
    for (int w = 32; w; w >>= 1) {
        acc += __shfl_down(acc, w, 64);
        if (METRIC == KM_COS) nrm += __shfl_down(nrm, w, 64);
    }
    if (lane) return;
    float dist;
    if (METRIC == KM_IP) dist = -acc;
    else if (METRIC == KM_COS) {
        const double denom = sqrt((double)nrm) * sqrt((double)qnorm);
        if (denom == 0.0) dist = 1.0f;
        else {
            double sim = (double)acc / denom;
            sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
            dist = (float)(1.0 - sim);
        }
    } else dist = acc;
    out[r] = dist;
}

void launch_pairwise(int metric, const float* rows, const float* query,
                     float qnorm, int64_t n, int dim, float* out,
                     hipStream_t stream) {
    if (!n) return;
    const int wpb = 4;
    const int64_t grid = (n + wpb - 1) / wpb;
    switch (metric) {
    case KM_L2SQ:
        hipLaunchKernelGGL((pairwise_kernel<KM_L2SQ>), dim3((uint32_t)grid),
                           dim3(wpb * 64), 0, stream, rows, query, qnorm, n,
                           dim, out);
        break;
    case KM_IP:
        hipLaunchKernelGGL((pairwise_kernel<KM_IP>), dim3((uint32_t)grid),
                           dim3(wpb * 64), 0, stream, rows, query, qnorm, n,
                           dim, out);
        break;
    case KM_COS:
        hipLaunchKernelGGL((pairwise_kernel<KM_COS>), dim3((uint32_t)grid),
                           dim3(wpb * 64), 0, stream, rows, query, qnorm, n,
                           dim, out);
        break;
    default:
        hipLaunchKernelGGL((pairwise_kernel<KM_L1>), dim3((uint32_t)grid),
                           dim3(wpb * 64), 0, stream, rows, query, qnorm, n,
                           dim, out);
        break;
    }
}

/* --------------------------- query norms ---------------------------------
 * one wave per query; f32 accumulate (matches distance_func.go cosine's f32
 * norm accumulation within the 1e-5 parity tolerance). */
__global__ void qnorm_kernel(const float* __restrict__ queries, int nq,
                             int dpad, float* __restrict__ qnorms) {
    const int q = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    if (q >= nq) return;
    const float* src = queries + (int64_t)q * dpad;
    float s = 0.f;
    for (int e = lane; e < dpad; e += 64) s += src[e] * src[e];
#pragma unroll
    for (int w = 32; w; w >>= 1) s += __shfl_down(s, w, 64);
    if (lane == 0) qnorms[q] = s;
}

void launch_qnorms(const float* queries, int nq, int dpad, float* qnorms,
                   hipStream_t stream) {
    if (!nq) return;
    const int wpb = 4;
    const int grid = (nq + wpb - 1) / wpb;
    hipLaunchKernelGGL(qnorm_kernel, dim3(grid), dim3(wpb * 64), 0, stream,
                       queries, nq, dpad, qnorms);
}

}  // namespace moann
