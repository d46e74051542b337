/* CDNA4 IVF-PQ search kernels (BASELINE.json config 5).
 *
 * ADC with per-subspace codebooks over coarse residuals (the published
 * IVF-PQ algorithm the reference drives through cuVS — oracle/pq.py holds
 * the CPU restatement these kernels are parity-tested against; the LUT-in-
 * LDS shape follows the cuVS precedent noted in SURVEY §8 f1, with f32
 * accumulation).
 *
 * One workgroup per (probed list, query) pair: build the per-(query,list)
 * LUT [nsub][256] f32 in LDS (residual against the list's coarse centroid),
 * then stream the list's packed codes — groups of 64 rows, subspace-major
 * uchar4 quads: codes[(g*nsub/4 + mq)*256 + lane*4 + c] = code of row
 * g*64+lane, subspace 4*mq+c — one coalesced 256 B read per wave step. */

#include <hip/hip_runtime.h>
#include <algorithm>
#include <cfloat>
#include <cstdint>

#include "moann_internal.h"

namespace moann {

/* ---- ADC table decomposition ------------------------------------------
 * ||(q - c_l)_m - b||^2 = A_q[m][b] + B_l[m][b] - 2*(q . c_l)  summed over m,
 * where A_q[m][b] = ||q_m - cb[m][b]||^2 (per QUERY, computed once per
 * batch) and B_l[m][b] = ||c_l,m||^2 + 2*(c_l,m . cb[m][b]) (per LIST,
 * precomputed at BUILD). The naive per-(query,list) LUT rebuild read the
 * whole codebook (nsub*256*sd floats) per job and was the measured
 * bottleneck (103 GB/s effective on config-5 shape); with the split the
 * per-job LUT phase is two coalesced nsub*256 streams + one dot. */

__global__ __launch_bounds__(256) void pq_btab_kernel(
    const float* __restrict__ cents, /* [nlist][dpad] */
    const float* __restrict__ cb,    /* [nsub][256][sd] */
    int nsub, int sd, int dpad, int64_t nlist,
    float* __restrict__ btab /* [nlist][nsub*256] */) {
    const int64_t total = nlist * nsub * 256;
    const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    const int e = (int)(idx % (nsub * 256));
    const int64_t l = idx / (nsub * 256);
    const int m = e >> 8, c = e & 255;
    const float* cm = cents + l * dpad + m * sd;
    const float* bm = cb + ((int64_t)m * 256 + c) * sd;
    float cc = 0.f, cbdot = 0.f;
    for (int t = 0; t < sd; ++t) {
        cc = fmaf(cm[t], cm[t], cc);
        cbdot = fmaf(cm[t], bm[t], cbdot);
    }
    btab[idx] = cc + 2.f * cbdot;
}

__global__ __launch_bounds__(256) void pq_atab_kernel(
    const float* __restrict__ queries, /* [nq][dpad] */
    const float* __restrict__ cb, int nsub, int sd, int dpad, int64_t nq,
    float* __restrict__ atab /* [nq][nsub*256] */) {
    const int64_t total = nq * nsub * 256;
    const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    const int e = (int)(idx % (nsub * 256));
    const int64_t q = idx / (nsub * 256);
    const int m = e >> 8, c = e & 255;
    const float* qm = queries + q * dpad + m * sd;
    const float* bm = cb + ((int64_t)m * 256 + c) * sd;
    float a = 0.f;
    for (int t = 0; t < sd; ++t) {
        const float r = qm[t] - bm[t];
        a = fmaf(r, r, a);
    }
    atab[idx] = a;
}

void launch_pq_tables(const float* cents, const float* queries,
                      const float* cb, int nsub, int sd, int dpad,
                      int64_t nlist_or_nq, bool btab_mode, float* out,
                      hipStream_t stream) {
    const int64_t total = nlist_or_nq * nsub * 256;
    if (!total) return;
    const int64_t grid = (total + 255) / 256;
    if (btab_mode)
        hipLaunchKernelGGL(pq_btab_kernel, dim3((uint32_t)grid), dim3(256), 0,
                           stream, cents, cb, nsub, sd, dpad, nlist_or_nq,
                           out);
    else
        hipLaunchKernelGGL(pq_atab_kernel, dim3((uint32_t)grid), dim3(256), 0,
                           stream, queries, cb, nsub, sd, dpad, nlist_or_nq,
                           out);
}

/* Per-row ADC constant, folded at BUILD: row_b[slot] = sum_m B_l[m][code_m]
 * — query-independent, so the search LUT is A_q ONLY and one workgroup
 * serves a whole QUERY (every probed list) with a single LDS table.
 * dist(q,row) = sum_m A_q[m][code_m] + row_b[slot] - 2*(q . c_l), and
 * -2(q.c_l) = cdist(q,l) - |q|^2 - |c_l|^2 comes free from the coarse-rank
 * stage's exact L2 distances. */
template <int W>
__global__ __launch_bounds__(256) void pq_rowb_kernel(
    const uint8_t* __restrict__ codes, const float* __restrict__ btab,
    const int32_t* __restrict__ g2list, const int64_t* __restrict__ g2slot,
    const int32_t* __restrict__ g2valid, int64_t ngroups, int nsub,
    float* __restrict__ row_b) {
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int64_t g = (int64_t)blockIdx.x * 4 + wave;
    if (g >= ngroups) return;
    const float* bl = btab + (int64_t)g2list[g] * nsub * 256;
    const uint8_t* gp = codes + g * (int64_t)nsub * 64;
    float acc = 0.f;
    if (W == 16) {
        const uint4* cptr = (const uint4*)gp + lane;
        for (int mq = 0; mq < (nsub >> 4); ++mq) {
            const uint4 cd = cptr[mq * 64];
            const uint32_t w[4] = {cd.x, cd.y, cd.z, cd.w};
            for (int b = 0; b < 4; ++b) {
                const int mbase = (16 * mq + 4 * b) * 256;
                acc += bl[mbase + (w[b] & 255)];
                acc += bl[mbase + 256 + ((w[b] >> 8) & 255)];
                acc += bl[mbase + 512 + ((w[b] >> 16) & 255)];
                acc += bl[mbase + 768 + (w[b] >> 24)];
            }
        }
    } else {
        const uchar4* cptr = (const uchar4*)gp + lane;
        for (int mq = 0; mq < (nsub >> 2); ++mq) {
            const uchar4 cd = cptr[mq * 64];
            acc += bl[(4 * mq + 0) * 256 + cd.x];
            acc += bl[(4 * mq + 1) * 256 + cd.y];
            acc += bl[(4 * mq + 2) * 256 + cd.z];
            acc += bl[(4 * mq + 3) * 256 + cd.w];
        }
    }
    if (lane < g2valid[g]) row_b[g2slot[g] + lane] = acc;
}

void launch_pq_rowb(bool w16, const uint8_t* codes, const float* btab,
                    const int32_t* g2list, const int64_t* g2slot,
                    const int32_t* g2valid, int64_t ngroups, int nsub,
                    float* row_b, hipStream_t stream) {
    if (!ngroups) return;
    const dim3 grid((uint32_t)((ngroups + 3) / 4)), block(256);
    if (w16)
        hipLaunchKernelGGL(pq_rowb_kernel<16>, grid, block, 0, stream, codes,
                           btab, g2list, g2slot, g2valid, ngroups, nsub,
                           row_b);
    else
        hipLaunchKernelGGL(pq_rowb_kernel<4>, grid, block, 0, stream, codes,
                           btab, g2list, g2slot, g2valid, ngroups, nsub,
                           row_b);
}

/* One workgroup per QUERY: A_q LUT once in LDS, then every probed list's
 * codes streamed (W=16: dwordx4 over the 16-byte-unit layout). */
template <int W>
__global__ __launch_bounds__(1024) void pq_scan_q_kernel(
    const uint8_t* __restrict__ codes, const float* __restrict__ atab,
    const float* __restrict__ row_b, const float* __restrict__ qnorms,
    const float* __restrict__ cnorm2,
    const int32_t* __restrict__ csel_slots,
    const float* __restrict__ csel_dists, int probe,
    const int64_t* __restrict__ qoffs, const int64_t* __restrict__ probe_offs,
    const int64_t* __restrict__ list_gbase,
    const int32_t* __restrict__ list_rows,
    const int64_t* __restrict__ list_slot_base,
    const uint32_t* __restrict__ filter_bitset, /* bit per GLOBAL slot;
        cleared bit excludes (same contract as the flat scans); null = none */
    int nsub, float* __restrict__ dists_out) {
    extern __shared__ float lut[]; /* [nsub*256] */
    const int q = blockIdx.x;
    const float* aq = atab + (int64_t)q * nsub * 256;
    for (int e = threadIdx.x; e < nsub * 256; e += blockDim.x)
        lut[e] = aq[e];
    __syncthreads();
    const float qn2 = qnorms[q];
    /* 1024 threads = 16 waves: the LUT caps residency at one workgroup
     * per CU, so the waves come from block WIDTH, not block count */
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int nwave = blockDim.x >> 6;

    for (int r = 0; r < probe; ++r) {
        const int l = csel_slots[q * probe + r];
        if (l < 0) continue;
        const int rows = list_rows[l];
        if (!rows) continue;
        const float dconst = csel_dists[q * probe + r] - qn2 - cnorm2[l];
        const int64_t ob = qoffs[q] + probe_offs[q * (probe + 1) + r];
        const int64_t baseg = list_gbase[l];
        const int64_t sbase = list_slot_base[l];
        const int lg = (rows + 63) >> 6;
        for (int g = wave; g < lg; g += nwave) {
            const uint8_t* gp = codes + (baseg + g) * (int64_t)nsub * 64;
            float acc = 0.f;
            if (W == 16) {
                const uint4* cptr = (const uint4*)gp + lane;
#pragma unroll 2
                for (int mq = 0; mq < (nsub >> 4); ++mq) {
                    const uint4 cd = cptr[mq * 64];
                    const uint32_t w[4] = {cd.x, cd.y, cd.z, cd.w};
#pragma unroll
                    for (int b = 0; b < 4; ++b) {
                        const int mbase = (16 * mq + 4 * b) * 256;
                        acc += lut[mbase + (w[b] & 255)];
                        acc += lut[mbase + 256 + ((w[b] >> 8) & 255)];
                        acc += lut[mbase + 512 + ((w[b] >> 16) & 255)];
                        acc += lut[mbase + 768 + (w[b] >> 24)];
                    }
                }
            } else {
                const uchar4* cptr = (const uchar4*)gp + lane;
#pragma unroll 2
                for (int mq = 0; mq < (nsub >> 2); ++mq) {
                    const uchar4 cd = cptr[mq * 64];
                    acc += lut[(4 * mq + 0) * 256 + cd.x];
                    acc += lut[(4 * mq + 1) * 256 + cd.y];
                    acc += lut[(4 * mq + 2) * 256 + cd.z];
                    acc += lut[(4 * mq + 3) * 256 + cd.w];
                }
            }
            const int row = g * 64 + lane;
            if (row < rows) {
                const int64_t gs = sbase + row;
                const bool pass =
                    !filter_bitset ||
                    ((filter_bitset[gs >> 5] >> (gs & 31)) & 1u);
                dists_out[ob + row] =
                    pass ? acc + row_b[gs] + dconst : FLT_MAX;
            }
        }
    }
}

void launch_pq_scan_q(const uint8_t* codes, const float* atab,
                      const float* row_b, const float* qnorms,
                      const float* cnorm2, const int32_t* csel_slots,
                      const float* csel_dists, int probe,
                      const int64_t* qoffs, const int64_t* probe_offs,
                      const int64_t* list_gbase, const int32_t* list_rows,
                      const int64_t* list_slot_base,
                      const uint32_t* filter_bitset, int nsub, int nq,
                      float* dists_out, hipStream_t stream) {
    if (!nq) return;
    const size_t shmem = (size_t)nsub * 256 * 4;
    if ((nsub & 15) == 0)
        hipLaunchKernelGGL(pq_scan_q_kernel<16>, dim3(nq), dim3(1024), shmem,
                           stream, codes, atab, row_b, qnorms, cnorm2,
                           csel_slots, csel_dists, probe, qoffs, probe_offs,
                           list_gbase, list_rows, list_slot_base,
                           filter_bitset, nsub, dists_out);
    else
        hipLaunchKernelGGL(pq_scan_q_kernel<4>, dim3(nq), dim3(1024), shmem,
                           stream, codes, atab, row_b, qnorms, cnorm2,
                           csel_slots, csel_dists, probe, qoffs, probe_offs,
                           list_gbase, list_rows, list_slot_base,
                           filter_bitset, nsub, dists_out);
}

/* pack codes: thread per output uchar4 */
__global__ void pq_pack_kernel(const uint8_t* __restrict__ codes_rowmajor,
                               int nsub,
                               const int64_t* __restrict__ group_rowbase,
                               const int32_t* __restrict__ group_valid,
                               const int64_t* __restrict__ slot_rows,
                               int64_t ngroups,
                               uint8_t* __restrict__ packed) {
    const int64_t nquads = ngroups * (nsub >> 2) * 64;
    const int64_t gstride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < nquads; idx += gstride) {
    const int lane = (int)(idx & 63);
    const int64_t rest = idx >> 6;
    const int mq = (int)(rest % (nsub >> 2));
    const int64_t g = rest / (nsub >> 2);
    uchar4 v = {0, 0, 0, 0};
    if (lane < group_valid[g]) {
        const int64_t row = slot_rows[group_rowbase[g] + lane];
        const uint8_t* src = codes_rowmajor + row * (int64_t)nsub + 4 * mq;
        v.x = src[0]; v.y = src[1]; v.z = src[2]; v.w = src[3];
    }
    ((uchar4*)packed)[idx] = v;
    }
}

void launch_pq_pack(const uint8_t* codes_rowmajor, int nsub,
                    const int64_t* group_rowbase, const int32_t* group_valid,
                    const int64_t* slot_rows, int64_t ngroups, uint8_t* packed,
                    hipStream_t stream) {
    const int64_t nquads = ngroups * (int64_t)(nsub >> 2) * 64;
    if (!nquads) return;
    const int block = 256;
    const int64_t grid = std::min<int64_t>((nquads + block - 1) / block,
                                           1 << 22);
    hipLaunchKernelGGL(pq_pack_kernel, dim3((uint32_t)grid), dim3(block), 0,
                       stream, codes_rowmajor, nsub, group_rowbase,
                       group_valid, slot_rows, ngroups, packed);
}

}  // namespace moann
