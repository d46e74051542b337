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

__global__ __launch_bounds__(256) void pq_scan_kernel(
    const uint8_t* __restrict__ codes, /* packed, see header comment */
    const float* __restrict__ queries, /* [nq][dpad] */
    const float* __restrict__ cents,   /* [nlist][dpad] row-major */
    const float* __restrict__ cb,      /* [nsub][256][sd] */
    int nsub, int sd, int dpad,
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gcount, const int32_t* __restrict__ j_rows,
    const int32_t* __restrict__ j_query, const int32_t* __restrict__ j_list,
    const int64_t* __restrict__ j_outbase,
    float* __restrict__ dists_out) {
    extern __shared__ float lut[]; /* [nsub][256] */
    const int j = blockIdx.x;
    const int q = j_query[j], l = j_list[j];
    const float* qv = queries + (int64_t)q * dpad;
    const float* cv = cents + (int64_t)l * dpad;

    /* LUT[m][c] = || (q - cent_l)_m - cb[m][c] ||^2, f32 sequential over
     * the sub-dims (oracle/pq.py accumulates the same way) */
    for (int e = threadIdx.x; e < nsub * 256; e += blockDim.x) {
        const int m = e >> 8, c = e & 255;
        const float* cbe = cb + ((int64_t)m * 256 + c) * sd;
        float s = 0.f;
        for (int t = 0; t < sd; ++t) {
            const int dimix = m * sd + t;
            const float r = qv[dimix] - cv[dimix] - cbe[t];
            s = fmaf(r, r, s);
        }
        lut[e] = s;
    }
    __syncthreads();

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];
    const int64_t ob = j_outbase[j];
    const int mq4 = nsub >> 2;

    for (int g = wave; g < gcount; g += 4) {
        const uchar4* cptr =
            (const uchar4*)(codes + (baseg + g) * (int64_t)nsub * 64) + lane;
        float acc = 0.f;
        for (int mq = 0; mq < mq4; ++mq) {
            const uchar4 cd = cptr[mq * 64];
            /* subspace order 4*mq..4*mq+3 — matches the oracle's m loop */
            acc += lut[(4 * mq + 0) * 256 + cd.x];
            acc += lut[(4 * mq + 1) * 256 + cd.y];
            acc += lut[(4 * mq + 2) * 256 + cd.z];
            acc += lut[(4 * mq + 3) * 256 + cd.w];
        }
        const int row = g * 64 + lane;
        if (row < rows) dists_out[ob + row] = acc;
    }
}

void launch_pq_scan(const uint8_t* codes, const float* queries,
                    const float* cents, const float* cb, int nsub, int sd,
                    int dpad, int njobs, const int64_t* j_databaseg,
                    const int32_t* j_gcount, const int32_t* j_rows,
                    const int32_t* j_query, const int32_t* j_list,
                    const int64_t* j_outbase, float* dists_out,
                    hipStream_t stream) {
    if (!njobs) return;
    const size_t shmem = (size_t)nsub * 256 * 4;
    hipLaunchKernelGGL(pq_scan_kernel, dim3(njobs), dim3(256), shmem, stream,
                       codes, queries, cents, cb, nsub, sd, dpad, j_databaseg,
                       j_gcount, j_rows, j_query, j_list, j_outbase,
                       dists_out);
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
