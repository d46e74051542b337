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

/* One workgroup per (probed list, query): LUT[m][c] = A_q + B_l summed in
 * LDS, the per-pair constant -2*(q . c_l) block-reduced once, then the
 * list's packed codes streamed — W=16 (dwordx4, layout16) when
 * nsub%16==0, else the uchar4 layout. */
template <int W>
__global__ __launch_bounds__(256) void pq_scan_kernel(
    const uint8_t* __restrict__ codes,
    const float* __restrict__ queries, /* [nq][dpad] */
    const float* __restrict__ cents,   /* [nlist][dpad] row-major */
    const float* __restrict__ atab,    /* [nq][nsub*256] */
    const float* __restrict__ btab,    /* [nlist][nsub*256] */
    int nsub, int dpad,
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gcount, const int32_t* __restrict__ j_rows,
    const int32_t* __restrict__ j_query, const int32_t* __restrict__ j_list,
    const int64_t* __restrict__ j_outbase,
    float* __restrict__ dists_out) {
    extern __shared__ float lut[]; /* [nsub*256] + red[64] + dconst */
    float* red = lut + nsub * 256;
    const int j = blockIdx.x;
    const int q = j_query[j], l = j_list[j];

    const float* aq = atab + (int64_t)q * nsub * 256;
    const float* bl = btab + (int64_t)l * nsub * 256;
    for (int e = threadIdx.x; e < nsub * 256; e += blockDim.x)
        lut[e] = aq[e] + bl[e];

    /* block dot q . c_l */
    {
        const float* qv = queries + (int64_t)q * dpad;
        const float* cv = cents + (int64_t)l * dpad;
        float part = 0.f;
        for (int t = threadIdx.x; t < dpad; t += blockDim.x)
            part = fmaf(qv[t], cv[t], part);
        for (int off = 32; off; off >>= 1)
            part += __shfl_down(part, off, 64);
        if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = part;
        __syncthreads();
        if (threadIdx.x == 0)
            red[32] = -2.f * (red[0] + red[1] + red[2] + red[3]);
    }
    __syncthreads();
    const float dconst = red[32];

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];
    const int64_t ob = j_outbase[j];

    for (int g = wave; g < gcount; g += 4) {
        const uint8_t* gp = codes + (baseg + g) * (int64_t)nsub * 64;
        float acc = 0.f;
        if (W == 16) {
            const uint4* cptr = (const uint4*)gp + lane;
            const int m16 = nsub >> 4;
#pragma unroll 2
            for (int mq = 0; mq < m16; ++mq) {
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
            const int mq4 = nsub >> 2;
#pragma unroll 2
            for (int mq = 0; mq < mq4; ++mq) {
                const uchar4 cd = cptr[mq * 64];
                acc += lut[(4 * mq + 0) * 256 + cd.x];
                acc += lut[(4 * mq + 1) * 256 + cd.y];
                acc += lut[(4 * mq + 2) * 256 + cd.z];
                acc += lut[(4 * mq + 3) * 256 + cd.w];
            }
        }
        const int row = g * 64 + lane;
        if (row < rows) dists_out[ob + row] = acc + dconst;
    }
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

void launch_pq_scan(const uint8_t* codes, const float* queries,
                    const float* cents, const float* atab, const float* btab,
                    int nsub, int dpad, int njobs,
                    const int64_t* j_databaseg,
                    const int32_t* j_gcount, const int32_t* j_rows,
                    const int32_t* j_query, const int32_t* j_list,
                    const int64_t* j_outbase, float* dists_out,
                    hipStream_t stream) {
    if (!njobs) return;
    const size_t shmem = ((size_t)nsub * 256 + 40) * 4;
    if ((nsub & 15) == 0)
        hipLaunchKernelGGL(pq_scan_kernel<16>, dim3(njobs), dim3(256), shmem,
                           stream, codes, queries, cents, atab, btab, nsub,
                           dpad, j_databaseg, j_gcount, j_rows, j_query,
                           j_list, j_outbase, dists_out);
    else
        hipLaunchKernelGGL(pq_scan_kernel<4>, dim3(njobs), dim3(256), shmem,
                           stream, codes, queries, cents, atab, btab, nsub,
                           dpad, j_databaseg, j_gcount, j_rows, j_query,
                           j_list, j_outbase, dists_out);
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
    const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= nquads) return;
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

void launch_pq_pack(const uint8_t* codes_rowmajor, int nsub,
                    const int64_t* group_rowbase, const int32_t* group_valid,
                    const int64_t* slot_rows, int64_t ngroups, uint8_t* packed,
                    hipStream_t stream) {
    const int64_t nquads = ngroups * (int64_t)(nsub >> 2) * 64;
    if (!nquads) return;
    const int block = 256;
    const int64_t grid = (nquads + block - 1) / block;
    hipLaunchKernelGGL(pq_pack_kernel, dim3((uint32_t)grid), dim3(block), 0,
                       stream, codes_rowmajor, nsub, group_rowbase,
                       group_valid, slot_rows, ngroups, packed);
}

}  // namespace moann
