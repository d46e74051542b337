/* ORACLE — test infrastructure ONLY.
 *
 * CPU restatement of MatrixOne's IVF-Flat / brute-force ANN search hot path
 * (the reference at /root/reference, commit mounted read-only). This file is
 * the parity checker for the HIP/CDNA4 product path in matrixone_amd/csrc.
 * It may be imported, called, linked or executed ONLY by:
 *   - tests/            (parity + known-answer tests)
 *   - __graft_entry__.smoke()
 *   - bench.py's cpu_baseline leg (kind "port")
 * The product path NEVER routes through this code; matrixone_amd fails
 * loudly if its HIP extension is missing.
 *
 * Reference algorithms restated here (file:line cites relative to
 * /root/reference):
 *   - distance kernels + MO conventions:
 *       pkg/vectorindex/metric/distance_func.go:60-96   (L2DistanceSq, 8-wide
 *         unroll with pairwise grouping (d0d0+d1d1)+(d2d2+d3d3)+... )
 *       pkg/vectorindex/metric/distance_func.go:174-207 (InnerProduct returns
 *         MINUS a.b — the MO convention)
 *       pkg/vectorindex/metric/distance_func.go:211-286 (CosineDistance:
 *         1 - dot/(|a||b|), 4-wide unroll, clamp to [-1,1], denom==0 -> 1.0)
 *       pkg/vectorindex/metric/distance_func.go:112-150 (L1Distance)
 *       pkg/vectorindex/metric/types.go:234-251 (DistanceTransformIvfflat:
 *         sqrt when the query's function is l2_distance but the index metric
 *         is l2sq; DistanceTransformHnsw: usearch IP is 1-a.b, MO is -a.b)
 *   - IVF probe:
 *       pkg/vectorindex/ivfflat/search.go:265-308 (rankCentroids/findCentroids:
 *         brute force over nlist centroids in the INDEX metric, top-probe)
 *       pkg/vectorindex/ivfflat/search.go:917-1005 (probe: scan entries of the
 *         selected lists, per-entry distance in the index metric, ascending
 *         top-k of `limit`)
 *       pkg/vectorindex/ivfflat/search.go:1062-1077 (scoreFromQuantized:
 *         raw /= mul^2 for int8/uint8 quantized storage, then
 *         DistanceTransformIvfflat)
 *   - brute force:
 *       pkg/vectorindex/brute_force/brute_force.go:220-509 (exact search over
 *         all rows, bounded ascending top-k)
 *   - result padding/merge conventions (GPU precedent):
 *       cgo/cuvs/helper.h:137-167 (cpu_topk_merge_sharded: unfilled slots are
 *         (-1, FLT_MAX); dense ascending top-k per query)
 *
 * Parity pinning: oracle outputs are checked against the reference's OWN
 * arithmetic — libusearch_c built from thirdparties/usearch-2.25.3.tar.gz
 * (SimSIMD 6.5.3 inside) in oracle/_ref — by tests/test_oracle_vs_usearch.py,
 * and against committed golden fixtures in tests/golden/.
 */

#include <stdint.h>
#include <stddef.h>
#include <math.h>
#include <float.h>
#include <string.h>
#include <stdlib.h>

#ifdef _OPENMP
#include <omp.h>
#endif

#define MOANN_EXPORT __attribute__((visibility("default")))

/* Metric enum — mirrors pkg/vectorindex/metric/types.go MetricType values
 * (metric_type.go: Metric_L2Distance=0, L2sq=1, IP=2, Cosine=3, L1=4). */
enum {
    MOANN_METRIC_L2 = 0,     /* sqrt(sum((a-b)^2)) — applied as a transform  */
    MOANN_METRIC_L2SQ = 1,   /* sum((a-b)^2)                                 */
    MOANN_METRIC_IP = 2,     /* -a.b  (MO convention, distance_func.go:207)  */
    MOANN_METRIC_COS = 3,    /* 1 - a.b/(|a||b|)                             */
    MOANN_METRIC_L1 = 4,     /* sum(|a-b|)                                   */
};

/* ---------------- distance kernels (distance_func.go order) -------------- */

MOANN_EXPORT float moann_oracle_l2sq_f32(const float* p, const float* q, int64_t n) {
    /* distance_func.go:60-96 — 8-wide unroll, pairwise grouping. */
    float sum = 0.0f;
    int64_t i = 0;
    for (; i + 8 <= n; i += 8) {
        float d0 = p[i] - q[i];
        float d1 = p[i + 1] - q[i + 1];
        float d2 = p[i + 2] - q[i + 2];
        float d3 = p[i + 3] - q[i + 3];
        float d4 = p[i + 4] - q[i + 4];
        float d5 = p[i + 5] - q[i + 5];
        float d6 = p[i + 6] - q[i + 6];
        float d7 = p[i + 7] - q[i + 7];
        sum += (d0 * d0 + d1 * d1) + (d2 * d2 + d3 * d3) +
               (d4 * d4 + d5 * d5) + (d6 * d6 + d7 * d7);
    }
    for (; i < n; i++) {
        float d = p[i] - q[i];
        sum += d * d;
    }
    return sum;
}

MOANN_EXPORT float moann_oracle_ip_f32(const float* p, const float* q, int64_t n) {
    /* distance_func.go:174-207 — returns MINUS the dot product. */
    float sum = 0.0f;
    int64_t i = 0;
    for (; i + 8 <= n; i += 8) {
        sum += p[i] * q[i] + p[i + 1] * q[i + 1] + p[i + 2] * q[i + 2] +
               p[i + 3] * q[i + 3] + p[i + 4] * q[i + 4] + p[i + 5] * q[i + 5] +
               p[i + 6] * q[i + 6] + p[i + 7] * q[i + 7];
    }
    for (; i < n; i++)
        sum += p[i] * q[i];
    return -sum;
}

MOANN_EXPORT float moann_oracle_cos_f32(const float* p, const float* q, int64_t n) {
    /* distance_func.go:211-286 — 4-wide unroll, one pass; clamp; denom 0 -> 1. */
    if (n == 0) return 0.0f;
    float dot = 0.0f, n1 = 0.0f, n2 = 0.0f;
    int64_t i = 0;
    for (; i + 4 <= n; i += 4) {
        dot += p[i] * q[i] + p[i + 1] * q[i + 1] + p[i + 2] * q[i + 2] + p[i + 3] * q[i + 3];
        n1 += p[i] * p[i] + p[i + 1] * p[i + 1] + p[i + 2] * p[i + 2] + p[i + 3] * p[i + 3];
        n2 += q[i] * q[i] + q[i + 1] * q[i + 1] + q[i + 2] * q[i + 2] + q[i + 3] * q[i + 3];
    }
    for (; i < n; i++) {
        dot += p[i] * q[i];
        n1 += p[i] * p[i];
        n2 += q[i] * q[i];
    }
    double denom = sqrt((double)n1) * sqrt((double)n2);
    if (denom == 0.0) return 1.0f;
    double sim = (double)dot / denom;
    if (sim > 1.0) sim = 1.0;
    else if (sim < -1.0) sim = -1.0;
    return (float)(1.0 - sim);
}

MOANN_EXPORT float moann_oracle_l1_f32(const float* p, const float* q, int64_t n) {
    /* distance_func.go:112-150 */
    float sum = 0.0f;
    int64_t i = 0;
    for (; i + 8 <= n; i += 8) {
        sum += fabsf(p[i] - q[i]);
        sum += fabsf(p[i + 1] - q[i + 1]);
        sum += fabsf(p[i + 2] - q[i + 2]);
        sum += fabsf(p[i + 3] - q[i + 3]);
        sum += fabsf(p[i + 4] - q[i + 4]);
        sum += fabsf(p[i + 5] - q[i + 5]);
        sum += fabsf(p[i + 6] - q[i + 6]);
        sum += fabsf(p[i + 7] - q[i + 7]);
    }
    for (; i < n; i++)
        sum += fabsf(p[i] - q[i]);
    return sum;
}

/* Raw metric distance in the INDEX metric space (no sqrt transform): the
 * Metric_L2Distance index is byte-identical to Metric_L2sqDistance and the
 * sqrt is a search-time transform (metric/types.go:121-133,245-251). */
static inline float metric_dist(int metric, const float* a, const float* b, int64_t d) {
    switch (metric) {
    case MOANN_METRIC_L2:
    case MOANN_METRIC_L2SQ: return moann_oracle_l2sq_f32(a, b, d);
    case MOANN_METRIC_IP:   return moann_oracle_ip_f32(a, b, d);
    case MOANN_METRIC_COS:  return moann_oracle_cos_f32(a, b, d);
    case MOANN_METRIC_L1:   return moann_oracle_l1_f32(a, b, d);
    default:                return FLT_MAX;
    }
}

/* scoreFromQuantized (ivfflat/search.go:1062-1077) + DistanceTransformIvfflat
 * (metric/types.go:245-251): raw /= mul^2, then sqrt when the original SQL
 * function is l2_distance over an l2sq index. orig_l2 selects that case. */
static inline double transform_score(double raw, int metric, int orig_l2, double quant_mul) {
    if (quant_mul != 0.0 && quant_mul != 1.0)
        raw /= quant_mul * quant_mul;
    if (orig_l2 && (metric == MOANN_METRIC_L2 || metric == MOANN_METRIC_L2SQ))
        return sqrt(raw);
    return raw;
}

/* ----------------- bounded ascending top-k (per query) ------------------- */
/* Max-heap of the current k best (smallest) distances — the shape of
 * vectorindex.FastMaxHeap (pkg/vectorindex/index.go:209-309). Ties broken by
 * smaller id first on output for determinism (the reference heap leaves tie
 * order unspecified; tests compare with tie tolerance). */

typedef struct {
    float dist;
    int64_t id;
} cand_t;

static inline int cand_less(cand_t a, cand_t b) {
    /* ordering for "worse": larger dist = worse; tie: larger id = worse */
    if (a.dist != b.dist) return a.dist < b.dist;
    return a.id < b.id;
}

static void heap_sift_down(cand_t* h, int64_t n, int64_t i) {
    for (;;) {
        int64_t l = 2 * i + 1, r = 2 * i + 2, m = i;
        if (l < n && cand_less(h[m], h[l])) m = l;
        if (r < n && cand_less(h[m], h[r])) m = r;
        if (m == i) break;
        cand_t t = h[i]; h[i] = h[m]; h[m] = t;
        i = m;
    }
}

/* push candidate into bounded max-heap of size cap; *n is current size */
static inline void heap_push_bounded(cand_t* h, int64_t* n, int64_t cap, cand_t c) {
    if (*n < cap) {
        h[*n] = c;
        int64_t i = (*n)++;
        while (i > 0) {
            int64_t p = (i - 1) / 2;
            if (cand_less(h[p], h[i])) { cand_t t = h[i]; h[i] = h[p]; h[p] = t; i = p; }
            else break;
        }
    } else if (cand_less(c, h[0])) {
        h[0] = c;
        heap_sift_down(h, cap, 0);
    }
}

static int cand_cmp_asc(const void* pa, const void* pb) {
    const cand_t* a = (const cand_t*)pa;
    const cand_t* b = (const cand_t*)pb;
    if (a->dist < b->dist) return -1;
    if (a->dist > b->dist) return 1;
    if (a->id < b->id) return -1;
    if (a->id > b->id) return 1;
    return 0;
}

/* heap -> dense ascending output, padded with (-1, FLT_MAX) sentinels
 * (cgo/cuvs/helper.h:137-167 contract). */
static void heap_emit(cand_t* h, int64_t n, int64_t k, int metric, int orig_l2,
                      double quant_mul, int64_t* out_ids, float* out_dists) {
    qsort(h, (size_t)n, sizeof(cand_t), cand_cmp_asc);
    for (int64_t j = 0; j < k; j++) {
        if (j < n) {
            out_ids[j] = h[j].id;
            out_dists[j] = (float)transform_score((double)h[j].dist, metric, orig_l2, quant_mul);
        } else {
            out_ids[j] = -1;
            out_dists[j] = FLT_MAX;
        }
    }
}

/* --------------------------- brute force search -------------------------- */
/* Restates pkg/vectorindex/brute_force/brute_force.go:336-509 (GoBruteForce
 * with FastMaxHeap) — exact scan of all rows, ascending top-k.
 * ids==NULL means identity ids 0..n-1. */
MOANN_EXPORT void moann_oracle_bruteforce(
    int metric, int orig_l2,
    const float* base, int64_t n, int64_t d, const int64_t* ids,
    const float* queries, int64_t nq, int64_t k,
    int64_t* out_ids /* nq*k */, float* out_dists /* nq*k */) {
#pragma omp parallel
    {
        cand_t* heap = (cand_t*)malloc(sizeof(cand_t) * (size_t)(k > 0 ? k : 1));
#pragma omp for schedule(dynamic, 1)
        for (int64_t q = 0; q < nq; q++) {
            const float* qv = queries + q * d;
            int64_t hn = 0;
            for (int64_t r = 0; r < n; r++) {
                cand_t c;
                c.dist = metric_dist(metric, base + r * d, qv, d);
                c.id = ids ? ids[r] : r;
                heap_push_bounded(heap, &hn, k, c);
            }
            heap_emit(heap, hn, k, metric, orig_l2, /*quant_mul*/ 1.0,
                      out_ids + q * k, out_dists + q * k);
        }
        free(heap);
    }
}

/* ------------------------------ IVF probe -------------------------------- */
/* Index layout (CSR by list — mirrors the entries hidden table grouped by
 * centroid_id, ivfflat/search.go:917-1005):
 *   centroids:     [nlist][d] f32
 *   list_offsets:  [nlist+1]   slot range of list i is [off[i], off[i+1])
 *   slot_rows:     [nslots]    row index into vecs for each slot
 *   vecs:          [n][d] f32  entry vectors (quantized domain if quantized)
 *   ids:           [n]         entry ids (PKs); NULL = identity
 *
 * Search (ivfflat/search.go:702,917-1005):
 *   1. rank centroids in the index metric, take top-probe (search.go:290-308)
 *   2. scan all entries of those lists, distance in the index metric
 *   3. ascending top-k of limit, then scoreFromQuantized transform
 *
 * out_probe (optional, nq*probe, -1-padded): the ranked centroid ids actually
 * probed, for cross-checking list selection against the product path. */
MOANN_EXPORT void moann_oracle_ivf_search(
    int metric, int orig_l2, double quant_mul,
    const float* centroids, int64_t nlist, int64_t d,
    const float* vecs, const int64_t* ids,
    const int64_t* list_offsets, const int64_t* slot_rows,
    const float* queries, int64_t nq,
    int64_t probe, int64_t k,
    int64_t* out_ids /* nq*k */, float* out_dists /* nq*k */,
    int64_t* out_probe /* nq*probe or NULL */) {
    if (probe > nlist) probe = nlist;
#pragma omp parallel
    {
        cand_t* cheap = (cand_t*)malloc(sizeof(cand_t) * (size_t)(probe > 0 ? probe : 1));
        cand_t* heap = (cand_t*)malloc(sizeof(cand_t) * (size_t)(k > 0 ? k : 1));
#pragma omp for schedule(dynamic, 1)
        for (int64_t q = 0; q < nq; q++) {
            const float* qv = queries + q * d;
            /* 1. rank centroids (brute force in the index metric) */
            int64_t cn = 0;
            for (int64_t c = 0; c < nlist; c++) {
                cand_t cd;
                cd.dist = metric_dist(metric, centroids + c * d, qv, d);
                cd.id = c;
                heap_push_bounded(cheap, &cn, probe, cd);
            }
            qsort(cheap, (size_t)cn, sizeof(cand_t), cand_cmp_asc);
            if (out_probe) {
                for (int64_t j = 0; j < probe; j++)
                    out_probe[q * probe + j] = (j < cn) ? cheap[j].id : -1;
            }
            /* 2+3. scan selected lists, bounded top-k */
            int64_t hn = 0;
            for (int64_t j = 0; j < cn; j++) {
                int64_t list = cheap[j].id;
                for (int64_t s = list_offsets[list]; s < list_offsets[list + 1]; s++) {
                    int64_t r = slot_rows[s];
                    cand_t cd;
                    cd.dist = metric_dist(metric, vecs + r * d, qv, d);
                    cd.id = ids ? ids[r] : r;
                    heap_push_bounded(heap, &hn, k, cd);
                }
            }
            heap_emit(heap, hn, k, metric, orig_l2, quant_mul,
                      out_ids + q * k, out_dists + q * k);
        }
        free(cheap);
        free(heap);
    }
}

/* Sharded top-k merge — restates cgo/cuvs/helper.h:137-167
 * (cpu_topk_merge_sharded): merge nshard dense per-shard top-k blocks into a
 * global ascending top-k; -1 neighbors are skipped; pad (-1, FLT_MAX). */
MOANN_EXPORT void moann_oracle_topk_merge(
    int64_t nshard, int64_t nq, int64_t k,
    const int64_t* shard_ids /* [nshard][nq][k] */,
    const float* shard_dists /* [nshard][nq][k] */,
    int64_t* out_ids /* nq*k */, float* out_dists /* nq*k */) {
#pragma omp parallel
    {
        cand_t* heap = (cand_t*)malloc(sizeof(cand_t) * (size_t)(k > 0 ? k : 1));
#pragma omp for schedule(static)
        for (int64_t q = 0; q < nq; q++) {
            int64_t hn = 0;
            for (int64_t s = 0; s < nshard; s++) {
                const int64_t* sid = shard_ids + (s * nq + q) * k;
                const float* sdist = shard_dists + (s * nq + q) * k;
                for (int64_t j = 0; j < k; j++) {
                    if (sid[j] == -1) continue;
                    cand_t c; c.dist = sdist[j]; c.id = sid[j];
                    heap_push_bounded(heap, &hn, k, c);
                }
            }
            /* already-transformed scores: no further transform */
            qsort(heap, (size_t)hn, sizeof(cand_t), cand_cmp_asc);
            for (int64_t j = 0; j < k; j++) {
                if (j < hn) { out_ids[q * k + j] = heap[j].id; out_dists[q * k + j] = heap[j].dist; }
                else { out_ids[q * k + j] = -1; out_dists[q * k + j] = FLT_MAX; }
            }
        }
        free(heap);
    }
}

MOANN_EXPORT int moann_oracle_num_threads(void) {
#ifdef _OPENMP
    return omp_get_max_threads();
#else
    return 1;
#endif
}
