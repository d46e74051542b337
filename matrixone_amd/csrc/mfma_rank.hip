/* MFMA f32 centroid-rank GEMM — the "query×centroid GEMM" stage of the IVF
 * probe on gfx950 matrix cores.
 *
 * Computes the B×nlist distance matrix via the expanded form
 * (|q|^2 - 2 q·c + |c|^2 for L2sq; -q·c for IP; 1 - q·c/(|q||c|) for cos)
 * with `v_mfma_f32_16x16x4_f32` — exact f32 fmaf-chain numerics at the f32
 * matrix rate (157 TF peak; cdna_hip_programming.md §3). The EXPANSION
 * changes L2 rounding vs the reference CPU kernel's diff-form
 * (distance_func.go:60-96), exactly as the reference's own GPU coarse
 * search does (cuVS fused L2 uses the expanded form), so this path is
 * opt-in (MOANN_RANK=mfma; the default diff-form scan is what the parity
 * suite pins bit-exactly against the oracle — see DESIGN.md's rank-stage
 * writeup). Ranking order differs only at FP ties.
 *
 * Round-2 geometry (the round-1 version read 4-dim fragments straight from
 * global: 3.8% MfmaUtil, L2-latency-bound): LDS-staged, double-buffered
 * K-slabs. One 256-thread block owns a 32(query)×32(centroid) macro tile;
 * its 4 waves compute the 2×2 arrangement of 16×16 MFMA tiles. Each K step
 * stages Q[32][64] and C[32][64] through LDS (float4-coalesced global
 * loads, padded row stride 69 so the 64-lane fragment reads hit the
 * optimal 2 addresses per bank), prefetching slab i+1 while MFMAs
 * consume slab i.
 * A fragment: lane l holds Q[row l&15][k0 + (l>>4)];
 * B fragment: lane l holds C[col l&15][k0 + (l>>4)];
 * C/D layout: lane l, reg r -> row (l>>4)*4+r, col l&15 (ISA §10 map). */

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>

#include "moann_internal.h"

namespace moann {

using f32x4 = __attribute__((ext_vector_type(4))) float;

#define KS 64          /* K slab (dims per LDS stage)      */
#define LSTRIDE 69     /* padded LDS row stride (f32): bank = (5r + k) % 32,
                          distinct across the 64-lane (row, koff) fragment
                          reads -- the optimal 2-way minimum              */

template <int METRIC>
__global__ __launch_bounds__(256) void rank_gemm_kernel(
    const float* __restrict__ queries,   /* [nq][dpad]    */
    const float* __restrict__ cents,     /* [nlist][dpad] */
    const float* __restrict__ qnorms,    /* [nq] |q|^2    */
    const float* __restrict__ cnorms,    /* [nlist] |c|^2 */
    int nq, int nlist, int dpad,
    float* __restrict__ out /* [nq][nlist] */) {
    __shared__ float ldsq[2][32 * LSTRIDE];
    __shared__ float ldsc[2][32 * LSTRIDE];

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int tiles_c = (nlist + 31) / 32;
    /* XCD-aware contiguous remap: consecutive jobs in one XCD share the
     * centroid macro-column, so the 32-row C slab re-reads are L2-served */
    int macro;
    {
        const int nwg = gridDim.x, bid = blockIdx.x;
        const int q8 = nwg >> 3, r8 = nwg & 7;
        const int xcd = bid & 7, idx = bid >> 3;
        macro = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8)
                + idx;
        if (macro >= nwg) macro = bid;
    }
    const int mrow = macro / tiles_c, mcol = macro % tiles_c;
    const int qbase = mrow * 32, cbase = mcol * 32;
    if (qbase >= nq || cbase >= nlist) return;

    /* stage-in indexing: 256 threads move 32 rows x 64 dims as float4s —
     * thread t loads row (t>>3), quad (t&7), i.e. 8 float4 = 32 dims per
     * row-half; two halves per row via the +32 offset. */
    const int srow = threadIdx.x >> 3;       /* 0..31 */
    const int squad = (threadIdx.x & 7) * 4; /* dim offset 0..28 */
    const int qsrc = qbase + srow < nq ? qbase + srow : nq - 1;
    const int csrc = cbase + srow < nlist ? cbase + srow : nlist - 1;

    auto stage = [&](int buf, int k0) {
        /* dpad is padded to a multiple of 4; KS=64 may overrun dpad's tail
         * — clamp by zero-filling (zeros are identity for the dot). */
        const float* qp = queries + (int64_t)qsrc * dpad;
        const float* cp = cents + (int64_t)csrc * dpad;
#pragma unroll
        for (int half = 0; half < 2; ++half) {
            const int kk = k0 + half * 32 + squad;
            float4 qv = {0, 0, 0, 0}, cv = {0, 0, 0, 0};
            if (kk + 3 < dpad) {
                qv = *(const float4*)(qp + kk);
                cv = *(const float4*)(cp + kk);
            } else {
                for (int e = 0; e < 4 && kk + e < dpad; ++e) {
                    ((float*)&qv)[e] = qp[kk + e];
                    ((float*)&cv)[e] = cp[kk + e];
                }
            }
            float* dq = &ldsq[buf][srow * LSTRIDE + half * 32 + squad];
            float* dc = &ldsc[buf][srow * LSTRIDE + half * 32 + squad];
            *(float4*)dq = qv;
            *(float4*)dc = cv;
        }
    };

    /* wave w computes 16x16 tile (w>>1, w&1) of the 32x32 macro */
    const int tq = (wave >> 1) * 16, tc = (wave & 1) * 16;
    const int arow = tq + (lane & 15);  /* A fragment row (query)    */
    const int brow = tc + (lane & 15);  /* B fragment row (centroid) */
    const int koff = lane >> 4;         /* 0..3                      */

    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    stage(0, 0);
    __syncthreads();
    for (int k0 = 0; k0 < dpad; k0 += KS) {
        const int buf = (k0 / KS) & 1;
        if (k0 + KS < dpad) stage(buf ^ 1, k0 + KS);
        const float* qa = &ldsq[buf][arow * LSTRIDE + koff];
        const float* cb = &ldsc[buf][brow * LSTRIDE + koff];
#pragma unroll
        for (int u = 0; u < KS / 4; ++u)
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(qa[4 * u], cb[4 * u],
                                                       acc, 0, 0, 0);
        __syncthreads();
    }

    /* epilogue: lane l, reg r -> query row (l>>4)*4+r, centroid col l&15 */
    const int ocol = cbase + tc + (lane & 15);
    if (ocol >= nlist) return;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int orow = qbase + tq + (lane >> 4) * 4 + r;
        if (orow >= nq) continue;
        const float dot = acc[r];
        float dist;
        if (METRIC == KM_IP) {
            dist = -dot;
        } else if (METRIC == KM_COS) {
            const double denom =
                sqrt((double)cnorms[ocol]) * sqrt((double)qnorms[orow]);
            if (denom == 0.0) dist = 1.0f;
            else {
                double sim = (double)dot / denom;
                sim = sim > 1.0 ? 1.0 : (sim < -1.0 ? -1.0 : sim);
                dist = (float)(1.0 - sim);
            }
        } else { /* KM_L2SQ: expanded form */
            dist = qnorms[orow] + cnorms[ocol] - 2.0f * dot;
            if (dist < 0.f) dist = 0.f;
        }
        out[(int64_t)orow * nlist + ocol] = dist;
    }
}

/* Returns false when the metric has no MFMA ranking path (L1 keeps the
 * diff-form scan). */
bool launch_rank_gemm(int metric, const float* queries, const float* cents,
                      const float* qnorms, const float* cnorms, int nq,
                      int nlist, int dpad, float* out, hipStream_t stream) {
    if (metric == KM_L1) return false;
    const int tiles = ((nq + 31) / 32) * ((nlist + 31) / 32);
    switch (metric) {
    case KM_IP:
        hipLaunchKernelGGL((rank_gemm_kernel<KM_IP>), dim3(tiles), dim3(256),
                           0, stream, queries, cents, qnorms, cnorms, nq,
                           nlist, dpad, out);
        break;
    case KM_COS:
        hipLaunchKernelGGL((rank_gemm_kernel<KM_COS>), dim3(tiles), dim3(256),
                           0, stream, queries, cents, qnorms, cnorms, nq,
                           nlist, dpad, out);
        break;
    default:
        hipLaunchKernelGGL((rank_gemm_kernel<KM_L2SQ>), dim3(tiles),
                           dim3(256), 0, stream, queries, cents, qnorms,
                           cnorms, nq, nlist, dpad, out);
        break;
    }
    return true;
}

}  // namespace moann
