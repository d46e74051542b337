/* MFMA f32 centroid-rank GEMM — the "query×centroid GEMM" stage of the IVF
 * probe on gfx950 matrix cores.
 *
 * Computes the B×nlist distance matrix via the expanded form
 * (|q|^2 - 2 q·c + |c|^2 for L2sq; -q·c for IP; 1 - q·c/(|q||c|) for cos)
 * with `v_mfma_f32_16x16x4_f32` — exact f32 fmaf-chain numerics at the f32
 * vector rate (157 TF peak; cdna_hip_programming.md §3). The EXPANSION
 * changes L2 rounding vs the reference CPU kernel's diff-form
 * (distance_func.go:60-96), exactly as the reference's own GPU coarse
 * search does (cuVS fused L2 uses the expanded form), so this path is
 * opt-in (MOANN_RANK=mfma; default stays the diff-form scan used by the
 * parity suite). Ranking order differs only at FP ties.
 *
 * Geometry: one wave per 16(query)x16(centroid) output tile, K-loop of 4
 * dims per MFMA, operands read straight from global (queries ~3 MB and
 * centroids ~12.6 MB are L2/LLC-resident after the first tile row).
 * A fragment: lane l holds Q[row l&15][k0 + (l>>4)];
 * B fragment: lane l holds C[col l&15][k0 + (l>>4)];
 * C/D layout: lane l, reg r -> row (l>>4)*4+r, col l&15 (ISA §10 map). */

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>

#include "moann_internal.h"

namespace moann {

using f32x4 = __attribute__((ext_vector_type(4))) float;

template <int METRIC>
__global__ __launch_bounds__(256) void rank_gemm_kernel(
    const float* __restrict__ queries,   /* [nq][dpad]    */
    const float* __restrict__ cents,     /* [nlist][dpad] */
    const float* __restrict__ qnorms,    /* [nq] |q|^2    */
    const float* __restrict__ cnorms,    /* [nlist] |c|^2 */
    int nq, int nlist, int dpad,
    float* __restrict__ out /* [nq][nlist] */) {
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    /* 4 waves per block arranged 2x2 over a 32x32 macro tile */
    const int tiles_c = (nlist + 31) / 32;
    const int macro = blockIdx.x;
    const int mrow = macro / tiles_c, mcol = macro % tiles_c;
    const int qbase = mrow * 32 + (wave >> 1) * 16;
    const int cbase = mcol * 32 + (wave & 1) * 16;
    if (qbase >= nq || cbase >= nlist) return;

    const int qrow = lane & 15, koff = lane >> 4;
    /* clamp edge rows/cols to a valid row; results discarded on store */
    const int qi = qbase + qrow < nq ? qbase + qrow : nq - 1;
    const int ci = cbase + qrow < nlist ? cbase + qrow : nlist - 1;
    const float* qptr = queries + (int64_t)qi * dpad + koff;
    const float* cptr = cents + (int64_t)ci * dpad + koff;

    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    int k0 = 0;
    /* 8-deep K unroll: 16 independent dword loads in flight per step */
    for (; k0 + 32 <= dpad; k0 += 32) {
        float a[8], b[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
            a[u] = qptr[k0 + 4 * u];
            b[u] = cptr[k0 + 4 * u];
        }
#pragma unroll
        for (int u = 0; u < 8; ++u)
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a[u], b[u], acc, 0, 0, 0);
    }
    for (; k0 < dpad; k0 += 4)
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(qptr[k0], cptr[k0], acc,
                                                   0, 0, 0);

    /* epilogue: lane l, reg r -> query row (l>>4)*4+r, centroid col l&15 */
    const int ocol = cbase + (lane & 15);
    if (ocol >= nlist) return;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int orow = qbase + (lane >> 4) * 4 + r;
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
