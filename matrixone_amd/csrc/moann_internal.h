/* Internal interfaces between moann_lib.cpp (host/ABI) and
 * moann_kernels.hip (CDNA4 device code). Not part of the public ABI. */

#ifndef MOANN_INTERNAL_H
#define MOANN_INTERNAL_H

#include <hip/hip_runtime.h>
#include <cstdint>

namespace moann {

/* Internal metric codes (template selectors for the scan kernel). */
enum MetricKind : int {
    KM_L2SQ = 0,  /* sum((x-q)^2)                     */
    KM_IP = 1,    /* -x.q (MO convention)             */
    KM_COS = 2,   /* 1 - x.q/(|x||q|), clamp, 0 -> 1  */
    KM_L1 = 3,    /* sum|x-q|                         */
};

/* One scan job: a (list segment, query tile) pair. SoA device arrays. */
struct ScanJobs {
    const int64_t* databaseg;  /* [njobs] base group index of the list       */
    const int32_t* gstart;     /* [njobs] first group (list-relative)        */
    const int32_t* gcount;     /* [njobs] groups this job scans              */
    const int32_t* rows;       /* [njobs] valid rows in the WHOLE list       */
    const int32_t* nq;         /* [njobs] queries in the tile (<= QT)        */
    const int32_t* qbase;      /* [njobs] offset into qslot arrays           */
    const int32_t* qslot_query;   /* [nqslots] query index                   */
    const int64_t* qslot_outbase; /* [nqslots] output base in dists buffer   */
    const int64_t* slot_base;     /* [njobs] global slot of list row 0 (for
                                     the membership filter; may be null)     */
    int njobs;
};

/* Launch the distance scan over interleaved packed data.
 * packed layout: group g, dim-quad j4, lane l, c -> packed[((g*dpad + 4*j4)*64) + l*4 + c]
 * i.e. [g][dpad/4][64][4] floats; row (g*64+l) holds vector dims 4*j4+c.
 * queries: [nq_total][dpad] device, zero-padded.
 * qnorms: [nq_total] sum-of-squares (cos only, else nullptr).
 * dists_out: ragged candidate buffer, written at qslot_outbase + row. */
void launch_scan(int metric, int qt, const float* packed, const float* queries,
                 const float* qnorms, int dpad, const ScanJobs& jobs,
                 float* dists_out, hipStream_t stream,
                 const uint32_t* filter_bitset = nullptr);

/* Hand-scheduled asm specialization of the scan for the flagship shape
 * (scan_asm768.hip): METRIC=L2SQ, dpad=768, QT=16, one group per wave.
 * launch_scan dispatches here itself when qt==16 fits; exposed for tests. */
void launch_scan_asm768(const float* packed, const float* queries,
                        int dpad, const ScanJobs& jobs, float* dists_out,
                        hipStream_t stream,
                        const uint32_t* filter_bitset = nullptr);

/* Per-query ascending top-k select (radix select + in-LDS bitonic sort).
 * Candidates for query q live at dists[off(q) .. off(q)+count(q)) where
 * off/count come from qoffs (ragged, [nq+1]) or uniform n (qoffs==nullptr).
 * out_slots: [nq][k] candidate index within the query's range, -1 padding.
 * out_dists: [nq][k] ascending, FLT_MAX padding. k <= 4096. */
void launch_topk(const float* dists, const int64_t* qoffs, int64_t uniform_n,
                 int nq, int k, int32_t* out_slots, float* out_dists,
                 hipStream_t stream);

/* Map selected candidate slots to entry ids + apply the score transform
 * (scoreFromQuantized: raw/mul^2, then sqrt for orig-l2 — ivfflat/search.go:
 * 1062-1077). probe_lists/-offs describe each query's probed lists in rank
 * order. list_slot_base maps list -> base slot of id_by_slot. */
void launch_gather(const int32_t* sel_slots, const float* sel_dists,
                   const int32_t* probe_lists, const int64_t* probe_offs,
                   const int64_t* list_slot_base, const int64_t* id_by_slot,
                   int nprobe, int nq, int k, int do_sqrt, double inv_mul2,
                   int64_t* out_ids, float* out_dists, hipStream_t stream);

/* Same transform for uniform (brute-force / centroid) results where the slot
 * IS the row: id = id_by_slot ? id_by_slot[slot] : slot. */
void launch_gather_uniform(const int32_t* sel_slots, const float* sel_dists,
                           const int64_t* id_by_slot, int nq, int k,
                           int do_sqrt, double inv_mul2,
                           int64_t* out_ids, float* out_dists,
                           hipStream_t stream);

/* Pack rows into the interleaved layout.
 * group_rowbase[g]: index into slot_rows of the group's first row;
 * group_valid[g]: valid rows in group g; slot_rows: slot -> source row. */
void launch_pack(const float* vecs, int dim, int dpad,
                 const int64_t* group_rowbase, const int32_t* group_valid,
                 const int64_t* slot_rows, int64_t ngroups, float* packed,
                 hipStream_t stream);

/* Inverse of pack (save_dir): packed -> row-major rows in slot order. */
void launch_unpack(const float* packed, int dim, int dpad,
                   const int64_t* group_slotbase, const int32_t* group_valid,
                   int64_t ngroups, int64_t slot_lo, int64_t slot_hi,
                   float* out, hipStream_t stream);

/* MFMA f32 centroid-rank GEMM (mfma_rank.hip). Returns false when the
 * metric has no MFMA path (L1). */
bool launch_rank_gemm(int metric, const float* queries, const float* cents,
                      const float* qnorms, const float* cnorms, int nq,
                      int nlist, int dpad, float* out, hipStream_t stream);

/* Quantized (int8/uint8) list scan + helpers. */
void launch_scan_i8(int metric, bool uns, const uint8_t* packed,
                    const uint8_t* queries_q, const int32_t* qnorms,
                    const int32_t* rownorms, int dpad,
                    const ScanJobs& jb, float* dists_out, hipStream_t stream,
                    const uint32_t* filter_bitset = nullptr, int qt = 8);
/* per-(group,lane) sum-of-squares over the packed byte rows (build-time,
 * feeds the dot-form scan's rn + qn - 2*dot L2) */
void launch_bytes_pack16(const uint8_t* rows_q, int dpad,
                         const int64_t* group_rowbase,
                         const int32_t* group_valid,
                         const int64_t* slot_rows, int64_t ngroups,
                         uint8_t* packed, hipStream_t stream);
void launch_rownorms_i8(bool uns, const uint8_t* packed, int64_t ngroups,
                        int dpad, int32_t* out, hipStream_t stream);
void launch_unpack_bytes(const uint8_t* packed, int dim, int dpad,
                         const int64_t* group_slotbase,
                         const int32_t* group_valid, int64_t ngroups,
                         int64_t slot_lo, int64_t slot_hi, uint8_t* out,
                         hipStream_t stream);
/* exact f32 re-rank of byte-stage top-R (two-stage scan; SQ8 + refine) */
void launch_refine(int metric, const float* rows_f32, const float* queries,
                   const float* qnorms, int dim, int dpad, int R, int nq,
                   const int32_t* rsel_slots, const float* rsel_dists,
                   const int32_t* probe_lists, const int64_t* probe_offs,
                   const int64_t* list_slot_base, int nprobe, float* refined,
                   hipStream_t stream);
void launch_compose_select(const int32_t* sel2, const int32_t* rsel_slots,
                           int R, int k, int nq, int32_t* out,
                           hipStream_t stream);
void launch_quantize_rows(bool uns, const float* in, int64_t nrows,
                          int in_stride, int dim, int dpad, float fmul,
                          float fadd, uint8_t* out, hipStream_t stream);
void launch_pq_pack(const uint8_t* codes_rowmajor, int nsub,
                    const int64_t* group_rowbase, const int32_t* group_valid,
                    const int64_t* slot_rows, int64_t ngroups, uint8_t* packed,
                    hipStream_t stream);
/* half (f16/bf16) storage: plain-cast quantization + decoded-f32 scan */
void launch_quantize_half_rows(bool bf, const float* in, int64_t nrows,
                               int in_stride, int dim, int dpad,
                               uint16_t* out, hipStream_t stream);
void launch_qnorms_h(bool bf, const uint16_t* q, int nq, int dpad,
                     float* out, hipStream_t stream);
void launch_scan_h(int metric, bool bf, const uint16_t* packed,
                   const uint16_t* queries_h, const float* qnorms, int dpad,
                   const ScanJobs& jb, float* dists_out, hipStream_t stream,
                   const uint32_t* filter_bitset = nullptr);
void launch_qnorms_i8(bool uns, const uint8_t* q, int nq, int dpad,
                      int32_t* out, hipStream_t stream);

/* 1xN pairwise distances (SQL builtin batch / mocl.cu counterpart). */
void launch_pairwise(int metric, const float* rows, const float* query,
                     float qnorm, int64_t n, int dim, float* out,
                     hipStream_t stream);

/* Sum-of-squares per query row (cosine). queries: [nq][dpad]. */
void launch_qnorms(const float* queries, int nq, int dpad, float* qnorms,
                   hipStream_t stream);

}  // namespace moann

#endif
