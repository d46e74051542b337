/* Hand-scheduled list-scan kernel for the flagship shape:
 * METRIC=L2SQ, dpad=768, QT=16 query tiles, one 64-row group per wave.
 *
 * Why it exists: at QT=16 register pressure hipcc refuses to software-
 * pipeline the d4 loop (it emits a full `s_waitcnt vmcnt(0)` drain per
 * iteration; profiles/r01_README.md records every source-level attempt,
 * including the sched_barrier ring that is still in scan_kernel<*,*,1>).
 * Here the entire 192-quad loop is ONE asm block that owns BOTH hardware
 * counters, so nothing the compiler does can break the schedule:
 *   - a 3-deep global-load ring in pinned VGPRs v[40:51], uniform
 *     `s_waitcnt vmcnt(2)` before each consumer => 2 cache-line loads
 *     always in flight per wave, across loop iterations;
 *   - double-buffered LDS query quads in v[52:59], `s_waitcnt lgkmcnt(1)`
 *     before each 8-op FMA cluster => one ds_read always in flight.
 * d4 = 192 quads = 63 full 3-rotations + 1 peeled no-reissue epilogue
 * (exactly 192 issues, so the ring never reads past the group).
 * All ring/buffer/temp/address registers are explicitly clobbered, so the
 * allocator cannot copy a pending load destination (the hazard that sank
 * the "+v"-operand ring variant: a copy of a not-yet-landed register is
 * garbage even though the counted wait itself was correct).
 *
 * Issue budget per 1 KiB consumed per wave: 128 fma-class VALU + ~40
 * overhead slots => ~14 TB/s chip-wide issue ceiling at 3 waves/SIMD,
 * comfortably above the ~6.3 TB/s HBM bound this kernel targets.
 *
 * Semantics are identical to scan_kernel<KM_L2SQ, 16, 1> in
 * moann_kernels.hip (same job arrays, same accumulation order — component
 * order inside a quad, quads ascending — so distances are bit-equal);
 * that generic kernel remains the path for every other (metric, dpad, QT).
 */

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>

#include "moann_internal.h"

namespace moann {

#define MO_A768_QT 16
#define MO_A768_DPAD 768

/* one query-tile step: wait for the landed qv read (one still in flight),
 * 4 sub + 4 fma against ring slot {X0..X3}, then TAIL (the next ds_read
 * issue, +address bump where due). Q regs alternate buf0 v[52:55] /
 * buf1 v[56:59] with t parity. */
#define MO_ST(ACC, Q0, Q1, Q2, Q3, X0, X1, X2, X3, TAIL)                   \
    "s_waitcnt lgkmcnt(1)\n\t"                                             \
    "v_sub_f32 v60, " X0 ", " Q0 "\n\t"                                    \
    "v_sub_f32 v61, " X1 ", " Q1 "\n\t"                                    \
    "v_sub_f32 v62, " X2 ", " Q2 "\n\t"                                    \
    "v_sub_f32 v63, " X3 ", " Q3 "\n\t"                                    \
    "v_fmac_f32 " ACC ", v60, v60\n\t"                                     \
    "v_fmac_f32 " ACC ", v61, v61\n\t"                                     \
    "v_fmac_f32 " ACC ", v62, v62\n\t"                                     \
    "v_fmac_f32 " ACC ", v63, v63\n\t"                                     \
    TAIL

#define MO_ST0(ACC, X0, X1, X2, X3, TAIL) \
    MO_ST(ACC, "v52", "v53", "v54", "v55", X0, X1, X2, X3, TAIL)
#define MO_ST1(ACC, X0, X1, X2, X3, TAIL) \
    MO_ST(ACC, "v56", "v57", "v58", "v59", X0, X1, X2, X3, TAIL)

/* one rotation: wait the ring (WAITC loads still in flight AFTER the wait),
 * 16 query-tile steps against ring slot X, then REISSUE (the next
 * global_load into X + 64-bit address bump — empty in the peeled epilogue).
 * ds_read offsets: step t issues read for tile t+2 of the CURRENT quad at
 * byte (t+2)*dpad*4 = (t+2)*3072; steps 14/15 issue tiles 0/1 of the NEXT
 * quad (v66, the quad base, advances by 16 after step 13's issue).
 * lgkm invariant: exactly 2 ds_reads outstanding at every step entry. */
#define MO_ROT(WAITC, X0, X1, X2, X3, REISSUE)                             \
    "s_waitcnt vmcnt(" WAITC ")\n\t"                                       \
    MO_ST0("%[a0]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:6144\n\t")                   \
    MO_ST1("%[a1]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:9216\n\t")                   \
    MO_ST0("%[a2]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:12288\n\t")                  \
    MO_ST1("%[a3]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:15360\n\t")                  \
    MO_ST0("%[a4]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:18432\n\t")                  \
    MO_ST1("%[a5]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:21504\n\t")                  \
    MO_ST0("%[a6]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:24576\n\t")                  \
    MO_ST1("%[a7]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:27648\n\t")                  \
    MO_ST0("%[a8]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:30720\n\t")                  \
    MO_ST1("%[a9]",  X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:33792\n\t")                  \
    MO_ST0("%[a10]", X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:36864\n\t")                  \
    MO_ST1("%[a11]", X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:39936\n\t")                  \
    MO_ST0("%[a12]", X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66 offset:43008\n\t")                  \
    MO_ST1("%[a13]", X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:46080\n\t"                   \
           "v_add_u32 v66, 16, v66\n\t")                                   \
    MO_ST0("%[a14]", X0, X1, X2, X3,                                       \
           "ds_read_b128 v[52:55], v66\n\t")                               \
    MO_ST1("%[a15]", X0, X1, X2, X3,                                       \
           "ds_read_b128 v[56:59], v66 offset:3072\n\t")                   \
    REISSUE

#define MO_REISSUE(XRANGE)                                                 \
    "global_load_dwordx4 " XRANGE ", v[64:65], off\n\t"                    \
    "v_add_co_u32 v64, vcc, 0x400, v64\n\t"                                \
    "v_addc_co_u32 v65, vcc, 0, v65, vcc\n\t"

__global__ __launch_bounds__(256) void scan_asm768_kernel(
    const float* __restrict__ packed, const float* __restrict__ queries,
    const float* __restrict__ qnorms, int dpad, /* must be 768 */
    const int64_t* __restrict__ j_databaseg,
    const int32_t* __restrict__ j_gstart, const int32_t* __restrict__ j_gcount,
    const int32_t* __restrict__ j_rows, const int32_t* __restrict__ j_nq,
    const int32_t* __restrict__ j_qbase,
    const int32_t* __restrict__ qslot_query,
    const int64_t* __restrict__ qslot_outbase,
    const int64_t* __restrict__ j_slot_base,
    const uint32_t* __restrict__ filter_bitset,
    float* __restrict__ dists_out) {
    (void)qnorms;
    constexpr int QT = MO_A768_QT;
    extern __shared__ float lds[]; /* [QT][768] (+[QT] unused norm slot) */
    float* ldsq = lds;

    /* XCD-aware bijective blockIdx remap — same as scan_kernel (T1) */
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
            const float* src = queries + (int64_t)q * MO_A768_DPAD;
            for (int e = threadIdx.x; e < MO_A768_DPAD; e += blockDim.x)
                ldsq[t * MO_A768_DPAD + e] = src[e];
        } else {
            for (int e = threadIdx.x; e < MO_A768_DPAD; e += blockDim.x)
                ldsq[t * MO_A768_DPAD + e] = 0.f;
        }
    }
    __syncthreads();

    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int gstart = j_gstart[j], gcount = j_gcount[j];
    const int rows = j_rows[j];
    const int64_t baseg = j_databaseg[j];

    /* LDS byte offset of tile 0, quad 0 (addrspacecast generic->local
     * yields the 32-bit LDS segment offset) */
    const uint32_t lbase = (uint32_t)(uintptr_t)(
        (__attribute__((address_space(3))) const float*)ldsq);

    for (int gp = gstart + wave; gp < gstart + gcount; gp += 4) {
        const float4* __restrict__ d0 =
            (const float4*)(packed + (baseg + gp) * (int64_t)64 *
                            MO_A768_DPAD) + lane;
        const uint64_t a64 = (uint64_t)(uintptr_t)d0;
        const uint32_t alo = (uint32_t)a64, ahi = (uint32_t)(a64 >> 32);

        float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f, a4 = 0.f, a5 = 0.f,
              a6 = 0.f, a7 = 0.f, a8 = 0.f, a9 = 0.f, a10 = 0.f, a11 = 0.f,
              a12 = 0.f, a13 = 0.f, a14 = 0.f, a15 = 0.f;
        int cnt = 63;

        asm volatile(
            /* drain compiler-issued traffic so counted waits start at 0 */
            "s_waitcnt vmcnt(0) lgkmcnt(0)\n\t"
            "v_mov_b32 v64, %[alo]\n\t"
            "v_mov_b32 v65, %[ahi]\n\t"
            "v_mov_b32 v66, %[lbase]\n\t"
            /* ring prologue: quads 0,1,2 in flight */
            "global_load_dwordx4 v[40:43], v[64:65], off\n\t"
            "v_add_co_u32 v64, vcc, 0x400, v64\n\t"
            "v_addc_co_u32 v65, vcc, 0, v65, vcc\n\t"
            "global_load_dwordx4 v[44:47], v[64:65], off\n\t"
            "v_add_co_u32 v64, vcc, 0x400, v64\n\t"
            "v_addc_co_u32 v65, vcc, 0, v65, vcc\n\t"
            "global_load_dwordx4 v[48:51], v[64:65], off\n\t"
            "v_add_co_u32 v64, vcc, 0x400, v64\n\t"
            "v_addc_co_u32 v65, vcc, 0, v65, vcc\n\t"
            /* qv prologue: tiles 0,1 of quad 0 in flight */
            "ds_read_b128 v[52:55], v66\n\t"
            "ds_read_b128 v[56:59], v66 offset:3072\n\t"
            "MOANN_L%=:\n\t"
            MO_ROT("2", "v40", "v41", "v42", "v43", MO_REISSUE("v[40:43]"))
            MO_ROT("2", "v44", "v45", "v46", "v47", MO_REISSUE("v[44:47]"))
            MO_ROT("2", "v48", "v49", "v50", "v51", MO_REISSUE("v[48:51]"))
            "s_sub_u32 %[cnt], %[cnt], 1\n\t"
            "s_cmp_lg_u32 %[cnt], 0\n\t"
            "s_cbranch_scc1 MOANN_L%=\n\t"
            /* peeled epilogue: quads 189..191, no reissue (192 issues total;
             * its steps 14/15 still prefetch LDS bytes <= 49152+16, inside
             * the (16*768+16)*4 B allocation — harmless, drained below) */
            MO_ROT("2", "v40", "v41", "v42", "v43", "")
            MO_ROT("1", "v44", "v45", "v46", "v47", "")
            MO_ROT("0", "v48", "v49", "v50", "v51", "")
            "s_waitcnt vmcnt(0) lgkmcnt(0)\n\t"
            : [a0] "+v"(a0), [a1] "+v"(a1), [a2] "+v"(a2), [a3] "+v"(a3),
              [a4] "+v"(a4), [a5] "+v"(a5), [a6] "+v"(a6), [a7] "+v"(a7),
              [a8] "+v"(a8), [a9] "+v"(a9), [a10] "+v"(a10), [a11] "+v"(a11),
              [a12] "+v"(a12), [a13] "+v"(a13), [a14] "+v"(a14),
              [a15] "+v"(a15), [cnt] "+s"(cnt)
            : [alo] "v"(alo), [ahi] "v"(ahi), [lbase] "v"(lbase)
            : "v40", "v41", "v42", "v43", "v44", "v45", "v46", "v47", "v48",
              "v49", "v50", "v51", "v52", "v53", "v54", "v55", "v56", "v57",
              "v58", "v59", "v60", "v61", "v62", "v63", "v64", "v65", "v66",
              "vcc", "scc", "memory");

        const float acc0[QT] = {a0, a1, a2,  a3,  a4,  a5,  a6,  a7,
                                a8, a9, a10, a11, a12, a13, a14, a15};

        const int row0 = gp * 64 + lane;
        bool pass0 = true;
        if (filter_bitset && row0 < rows) {
            const int64_t gs = j_slot_base[j] + row0;
            pass0 = (filter_bitset[gs >> 5] >> (gs & 31)) & 1u;
        }
        if (row0 < rows) {
#pragma unroll
            for (int t = 0; t < QT; ++t) {
                if (t >= nq) break;
                const int64_t ob = qslot_outbase[qbase + t];
                dists_out[ob + row0] = pass0 ? acc0[t] : FLT_MAX;
            }
        }
    }
}

void launch_scan_asm768(const float* packed, const float* queries,
                        int dpad, const ScanJobs& jb, float* dists_out,
                        hipStream_t stream, const uint32_t* filter_bitset) {
    const dim3 grid(jb.njobs), block(256);
    hipLaunchKernelGGL(scan_asm768_kernel, grid, block,
                       (MO_A768_QT * dpad + MO_A768_QT) * sizeof(float),
                       stream, packed, queries, nullptr, dpad, jb.databaseg,
                       jb.gstart, jb.gcount, jb.rows, jb.nq, jb.qbase,
                       jb.qslot_query, jb.qslot_outbase, jb.slot_base,
                       filter_bitset, dists_out);
}

}  // namespace moann
