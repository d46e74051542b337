/* libmoann_hip — IVF-PQ host side (include/moann.h PQ
 * section). Mirrors the lifecycle of the reference's gpu_ivf_pq_* ABI
 * (cgo/cuvs/ivf_pq_c.h shape) with the moann ingestion model: the harness
 * trains coarse centroids + PQ codebooks and supplies codes + assignments;
 * the library owns the HBM layout and the search pipeline:
 *   centroid rank (interleaved f32 scan, same kernel as IVF-Flat)
 *   -> per-(query,list) LUT-in-LDS ADC code scan (pq_kernels.hip)
 *   -> radix top-k -> id gather (shared kernels).
 * Results come back through the shared gpu_ivf_flat_get_* readout. */

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cfloat>
#include <cstdint>
#include <cstdlib>
#include <map>
#include <memory>
#include <mutex>
#include <numeric>
#include <vector>

#include "../../include/moann.h"
#include "moann_host_common.h"
#include "moann_internal.h"

namespace moann {
void launch_pq_scan_q(const uint8_t* codes, const float* atab,
                      const float* row_b, const float* qnorms,
                      const float* cnorm2, const int32_t* csel_slots,
                      const float* csel_dists, int probe,
                      const int64_t* qoffs, const int64_t* probe_offs,
                      const int64_t* list_gbase, const int32_t* list_rows,
                      const int64_t* list_slot_base,
                      const uint32_t* filter_bitset, int nsub, int nq,
                      float* dists_out, hipStream_t stream);
void launch_pq_rowb(bool w16, const uint8_t* codes, const float* btab,
                    const int32_t* g2list, const int64_t* g2slot,
                    const int32_t* g2valid, int64_t ngroups, int nsub,
                    float* row_b, hipStream_t stream);
void launch_pq_tables(const float* cents, const float* queries,
                      const float* cb, int nsub, int sd, int dpad,
                      int64_t nlist_or_nq, bool btab_mode, float* out,
                      hipStream_t stream);
void launch_pq_pack(const uint8_t* codes_rowmajor, int nsub,
                    const int64_t* group_rowbase, const int32_t* group_valid,
                    const int64_t* slot_rows, int64_t ngroups, uint8_t* packed,
                    hipStream_t stream);
}

namespace {

using namespace moann;

struct PqIndex {
    int device = 0;
    uint32_t dim = 0, dpad = 0, nlist = 0, nsub = 0, sd = 0;
    uint64_t cap = 0, count = 0;
    bool built = false;

    hipStream_t stream = nullptr;

    std::vector<int64_t> h_ids;
    std::vector<int32_t> h_assign;
    std::vector<float> h_centroids;
    std::vector<uint8_t> h_codes; /* staging, row-major [count][nsub] */
    std::vector<float> h_codebooks;
    std::vector<int64_t> h_id_by_slot; /* slot -> id (persistence) */

    uint8_t* d_codes = nullptr;       /* packed groups */
    float* d_cents_rows = nullptr;    /* [nlist][dpad] */
    float* d_cent_packed = nullptr;   /* interleaved (rank stage) */
    float* d_codebooks = nullptr;     /* [nsub][256][sd] */
    float* d_btab = nullptr;          /* [nlist][nsub*256] (build only) */
    float* d_rowb = nullptr;          /* [count] per-row ADC constant */
    float* d_cnorm2 = nullptr;        /* [nlist] |c_l|^2 */
    int64_t* d_gbase_dev = nullptr;   /* [nlist] list_gbase */
    int32_t* d_rows_dev = nullptr;    /* [nlist] list_rows */
    int64_t* d_id_by_slot = nullptr;
    int64_t* d_list_slot_base = nullptr;
    std::vector<int64_t> list_gbase;
    std::vector<int32_t> list_rows;
    std::vector<int64_t> list_slot_base;
    int64_t total_groups = 0, cent_groups = 0;

    MoannDevBuf w_queries, w_cdists, w_csel_slots, w_csel_dists, w_jobs;
    MoannDevBuf w_cand, w_sel_slots, w_sel_dists, w_probe_lists, w_probe_offs;
    MoannDevBuf w_qoffs, w_out_ids, w_out_dists, w_atab, w_qn, w_filter;

    std::mutex mu;
    moann_perf_t perf {};
    hipEvent_t ev[4] {};
    bool ev_made = false;

    ~PqIndex() {
        (void)hipSetDevice(device);
        for (auto p : {(void*)d_codes, (void*)d_cents_rows,
                       (void*)d_cent_packed, (void*)d_codebooks,
                       (void*)d_btab, (void*)d_rowb, (void*)d_cnorm2,
                       (void*)d_gbase_dev, (void*)d_rows_dev,
                       (void*)d_id_by_slot, (void*)d_list_slot_base})
            if (p) (void)hipFree(p);
        if (ev_made)
            for (auto& e : ev) (void)hipEventDestroy(e);
        if (stream) (void)hipStreamDestroy(stream);
    }
};

PqIndex* PX(moann_ivf_pq_c h) { return (PqIndex*)h; }

std::unique_ptr<MoannResult> pq_search(PqIndex* ix, const float* queries,
                                       bool on_device, uint64_t nq,
                                       uint32_t limit, uint32_t n_probes,
                                       const uint32_t* filter_words
                                       = nullptr) {
    std::lock_guard<std::mutex> lk(ix->mu);
    MOANN_HIP_CHECK(hipSetDevice(ix->device));
    auto res = std::make_unique<MoannResult>();
    res->neighbors.assign((size_t)nq * limit, -1);
    res->distances.assign((size_t)nq * limit, FLT_MAX);
    if (!nq) return res;
    if (!ix->built) throw std::runtime_error("index not built");
    if (ix->count == 0) return res;
    if (limit > 4096) throw std::runtime_error("limit > 4096 unsupported");
    uint32_t probe = std::min<uint32_t>(n_probes ? n_probes : 1, ix->nlist);
    const int dpad = (int)ix->dpad;
    const hipStream_t s = ix->stream;
    if (!ix->ev_made) {
        for (auto& e : ix->ev) MOANN_HIP_CHECK(hipEventCreate(&e));
        ix->ev_made = true;
    }

    /* queries padded to device */
    ix->w_queries.ensure((size_t)nq * dpad * 4);
    if ((int)ix->dim != dpad)
        MOANN_HIP_CHECK(hipMemsetAsync(ix->w_queries.ptr, 0,
                                       (size_t)nq * dpad * 4, s));
    MOANN_HIP_CHECK(hipMemcpy2DAsync(
        ix->w_queries.ptr, (size_t)dpad * 4, queries, (size_t)ix->dim * 4,
        (size_t)ix->dim * 4, nq,
        on_device ? hipMemcpyDeviceToDevice : hipMemcpyHostToDevice, s));
    const float* d_q = ix->w_queries.as<float>();

    /* 1. coarse rank: interleaved scan over the centroid matrix (shared
     * kernel; f32 L2sq — config-5 is L2) + top-probe */
    {
        ix->w_cdists.ensure((size_t)nq * ix->nlist * 4);
        const int qt = 8;
        const int ntiles = (int)((nq + qt - 1) / qt);
        const int64_t cgroups = ix->cent_groups;
        int64_t want_chunks = std::min<int64_t>(
            cgroups, std::max<int64_t>(1, (1024 + ntiles - 1) / ntiles));
        const int chunk = (int)((cgroups + want_chunks - 1) / want_chunks);
        const int nchunk = (int)((cgroups + chunk - 1) / chunk);
        const int njobs = ntiles * nchunk;
        const int nqslots = ntiles * qt;
        std::vector<int64_t> databaseg(njobs), qslot_outbase(nqslots);
        std::vector<int32_t> gstart(njobs), gcount(njobs), rows(njobs),
            jnq(njobs), qbase(njobs), qslot_query(nqslots);
        for (int t = 0; t < ntiles; ++t) {
            const int q0 = t * qt;
            const int tn = (int)std::min<int64_t>(qt, (int64_t)nq - q0);
            for (int u = 0; u < qt; ++u) {
                const int qi = std::min<int>(q0 + u, (int)nq - 1);
                qslot_query[t * qt + u] = qi;
                qslot_outbase[t * qt + u] = (int64_t)qi * ix->nlist;
            }
            for (int c = 0; c < nchunk; ++c) {
                const int jid = t * nchunk + c;
                databaseg[jid] = 0;
                gstart[jid] = (int32_t)(c * chunk);
                gcount[jid] =
                    (int32_t)std::min<int64_t>(chunk, cgroups - c * chunk);
                rows[jid] = (int32_t)ix->nlist;
                jnq[jid] = tn;
                qbase[jid] = t * qt;
            }
        }
        const size_t b64 = 8, b32 = 4;
        auto al8 = [](size_t b) { return (b + 7) & ~7ull; };
        ix->w_jobs.ensure(al8(njobs * b64) + 5 * al8(njobs * b32) +
                          al8(nqslots * b32) + al8(nqslots * b64) + 64);
        uint8_t* p = ix->w_jobs.as<uint8_t>();
        auto up = [&](const void* src, size_t bytes) {
            void* r = p;
            if (bytes)
                MOANN_HIP_CHECK(hipMemcpyAsync(p, src, bytes,
                                               hipMemcpyHostToDevice, s));
            p += (bytes + 7) & ~7ull;
            return r;
        };
        ScanJobs jb;
        jb.databaseg = (int64_t*)up(databaseg.data(), njobs * b64);
        jb.gstart = (int32_t*)up(gstart.data(), njobs * b32);
        jb.gcount = (int32_t*)up(gcount.data(), njobs * b32);
        jb.rows = (int32_t*)up(rows.data(), njobs * b32);
        jb.nq = (int32_t*)up(jnq.data(), njobs * b32);
        jb.qbase = (int32_t*)up(qbase.data(), njobs * b32);
        jb.qslot_query = (int32_t*)up(qslot_query.data(), nqslots * b32);
        jb.qslot_outbase = (int64_t*)up(qslot_outbase.data(), nqslots * b64);
        jb.slot_base = nullptr;
        jb.njobs = njobs;
        launch_scan(KM_L2SQ, 8, ix->d_cent_packed, d_q, nullptr, dpad, jb,
                    ix->w_cdists.as<float>(), s);
        MOANN_KCHECK("pq-rank-scan");
        ix->w_csel_slots.ensure((size_t)nq * probe * 4);
        ix->w_csel_dists.ensure((size_t)nq * probe * 4);
        launch_topk(ix->w_cdists.as<float>(), nullptr, ix->nlist, (int)nq,
                    (int)probe, ix->w_csel_slots.as<int32_t>(),
                    ix->w_csel_dists.as<float>(), s);
        MOANN_KCHECK("pq-rank-topk");
    }

    /* 2. D2H probe; (list, query) jobs for the code scan */
    std::vector<int32_t> h_probe((size_t)nq * probe);
    MOANN_HIP_CHECK(hipMemcpyAsync(h_probe.data(), ix->w_csel_slots.ptr,
                                   h_probe.size() * 4, hipMemcpyDeviceToHost,
                                   s));
    MOANN_HIP_CHECK(hipStreamSynchronize(s));
    MOANN_HIP_CHECK(hipGetLastError());

    std::vector<int64_t> probe_offs((size_t)nq * (probe + 1));
    std::vector<int64_t> qoffs(nq + 1, 0);
    for (uint64_t q = 0; q < nq; ++q) {
        int64_t acc = 0;
        probe_offs[q * (probe + 1)] = 0;
        for (uint32_t r = 0; r < probe; ++r) {
            const int32_t l = h_probe[q * probe + r];
            if (l >= 0) acc += ix->list_rows[l];
            probe_offs[q * (probe + 1) + r + 1] = acc;
        }
        qoffs[q + 1] = qoffs[q] + acc;
    }
    const int64_t total_cand = qoffs[nq];

    ix->w_cand.ensure(std::max<int64_t>(1, total_cand) * 4);
    /* offsets to device BEFORE the scan (the per-query kernel walks them) */
    ix->w_qoffs.ensure((nq + 1) * 8);
    MOANN_HIP_CHECK(hipMemcpyAsync(ix->w_qoffs.ptr, qoffs.data(),
                                   (nq + 1) * 8, hipMemcpyHostToDevice, s));
    ix->w_probe_lists.ensure((size_t)nq * probe * 4);
    ix->w_probe_offs.ensure(probe_offs.size() * 8);
    MOANN_HIP_CHECK(hipMemcpyAsync(ix->w_probe_lists.ptr, h_probe.data(),
                                   h_probe.size() * 4, hipMemcpyHostToDevice,
                                   s));
    MOANN_HIP_CHECK(hipMemcpyAsync(ix->w_probe_offs.ptr, probe_offs.data(),
                                   probe_offs.size() * 8,
                                   hipMemcpyHostToDevice, s));
    {
        /* per-query A table + |q|^2 for this batch (row_b / |c|^2 are
         * build-time; -2(q.c_l) falls out of the rank stage's distances) */
        ix->w_atab.ensure((size_t)nq * ix->nsub * 256 * 4);
        launch_pq_tables(nullptr, d_q, ix->d_codebooks, (int)ix->nsub,
                         (int)ix->sd, dpad, (int64_t)nq, false,
                         ix->w_atab.as<float>(), s);
        MOANN_KCHECK("pq-atab");
        ix->w_qn.ensure(nq * 4);
        launch_qnorms(d_q, (int)nq, dpad, ix->w_qn.as<float>(), s);
        MOANN_KCHECK("pq-qnorms");
        const uint32_t* d_filter = nullptr;
        if (filter_words) {
            const size_t fw = (size_t)((ix->count + 31) / 32);
            ix->w_filter.ensure(fw * 4);
            MOANN_HIP_CHECK(hipMemcpyAsync(ix->w_filter.ptr, filter_words,
                                           fw * 4, hipMemcpyHostToDevice,
                                           s));
            d_filter = ix->w_filter.as<uint32_t>();
        }
        MOANN_HIP_CHECK(hipEventRecord(ix->ev[0], s));
        launch_pq_scan_q(ix->d_codes, ix->w_atab.as<float>(), ix->d_rowb,
                         ix->w_qn.as<float>(), ix->d_cnorm2,
                         ix->w_csel_slots.as<int32_t>(),
                         ix->w_csel_dists.as<float>(), (int)probe,
                         ix->w_qoffs.as<int64_t>(),
                         ix->w_probe_offs.as<int64_t>(), ix->d_gbase_dev,
                         ix->d_rows_dev, ix->d_list_slot_base, d_filter,
                         (int)ix->nsub, (int)nq, ix->w_cand.as<float>(), s);
        MOANN_KCHECK("pq-scan");
        MOANN_HIP_CHECK(hipEventRecord(ix->ev[1], s));
    }

    /* 3. top-k + gather (shared kernels) */
    ix->w_sel_slots.ensure((size_t)nq * limit * 4);
    ix->w_sel_dists.ensure((size_t)nq * limit * 4);
    launch_topk(ix->w_cand.as<float>(), ix->w_qoffs.as<int64_t>(), 0, (int)nq,
                (int)limit, ix->w_sel_slots.as<int32_t>(),
                ix->w_sel_dists.as<float>(), s);
    MOANN_KCHECK("pq-cand-topk");
    ix->w_out_ids.ensure((size_t)nq * limit * 8);
    ix->w_out_dists.ensure((size_t)nq * limit * 4);
    launch_gather(ix->w_sel_slots.as<int32_t>(), ix->w_sel_dists.as<float>(),
                  ix->w_probe_lists.as<int32_t>(),
                  ix->w_probe_offs.as<int64_t>(), ix->d_list_slot_base,
                  ix->d_id_by_slot, (int)probe, (int)nq, (int)limit,
                  /*do_sqrt=*/0, 1.0, ix->w_out_ids.as<int64_t>(),
                  ix->w_out_dists.as<float>(), s);
    MOANN_KCHECK("pq-gather");
    MOANN_HIP_CHECK(hipMemcpyAsync(res->neighbors.data(), ix->w_out_ids.ptr,
                                   (size_t)nq * limit * 8,
                                   hipMemcpyDeviceToHost, s));
    MOANN_HIP_CHECK(hipMemcpyAsync(res->distances.data(), ix->w_out_dists.ptr,
                                   (size_t)nq * limit * 4,
                                   hipMemcpyDeviceToHost, s));
    MOANN_HIP_CHECK(hipStreamSynchronize(s));
    MOANN_HIP_CHECK(hipGetLastError());

    float ms = 0;
    MOANN_HIP_CHECK(hipEventElapsedTime(&ms, ix->ev[0], ix->ev[1]));
    ix->perf.scan_launches++;
    ix->perf.scan_ms += ms;
    /* algorithmic bytes: candidate codes (nsub B each) + per-job LUT build */
    ix->perf.scan_bytes += (uint64_t)total_cand * ix->nsub;
    ix->perf.scan_rows += (uint64_t)total_cand;
    return res;
}

}  // namespace

extern "C" {

moann_ivf_pq_c moann_ivf_pq_new(uint32_t dimension, uint32_t n_sub,
                                uint32_t n_lists, uint64_t total_count,
                                const int* devices, int device_count,
                                void* errmsg) {
    try {
        if (!n_sub || dimension % n_sub)
            throw std::runtime_error("dimension must divide by n_sub");
        if (n_sub % 4)
            throw std::runtime_error("n_sub must be a multiple of 4");
        auto ix = std::make_unique<PqIndex>();
        ix->device = (devices && device_count > 0) ? devices[0] : 0;
        MOANN_HIP_CHECK(hipSetDevice(ix->device));
        MOANN_HIP_CHECK(hipStreamCreate(&ix->stream));
        ix->dim = dimension;
        ix->dpad = (dimension + 3) & ~3u;
        ix->nsub = n_sub;
        ix->sd = dimension / n_sub;
        ix->nlist = n_lists;
        ix->cap = total_count;
        ix->h_codes.reserve((size_t)total_count * n_sub);
        ix->h_ids.reserve(total_count);
        return (moann_ivf_pq_c)ix.release();
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_new", e.what());
        return nullptr;
    }
}

void moann_ivf_pq_set_centroids(moann_ivf_pq_c h, const float* centroids,
                                uint32_t n_lists, void* errmsg) {
    try {
        auto ix = PX(h);
        if (n_lists != ix->nlist)
            throw std::runtime_error("number of centroids != n_lists");
        ix->h_centroids.assign(centroids,
                               centroids + (size_t)n_lists * ix->dim);
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_set_centroids", e.what());
    }
}

void moann_ivf_pq_set_codebooks(moann_ivf_pq_c h, const float* codebooks,
                                void* errmsg) {
    try {
        auto ix = PX(h);
        ix->h_codebooks.assign(codebooks, codebooks + (size_t)ix->nsub * 256 *
                                                          ix->sd);
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_set_codebooks", e.what());
    }
}

void moann_ivf_pq_add_codes(moann_ivf_pq_c h, const uint8_t* codes,
                            uint64_t n, const int64_t* ids, void* errmsg) {
    try {
        auto ix = PX(h);
        if (ix->built) throw std::runtime_error("index already built");
        if (ix->count + n > ix->cap)
            throw std::runtime_error("exceeds capacity");
        ix->h_codes.insert(ix->h_codes.end(), codes,
                           codes + (size_t)n * ix->nsub);
        for (uint64_t i = 0; i < n; ++i)
            ix->h_ids.push_back(ids ? ids[i] : (int64_t)(ix->count + i));
        ix->count += n;
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_add_codes", e.what());
    }
}

void moann_ivf_pq_set_assignments(moann_ivf_pq_c h,
                                  const int32_t* assignments, uint64_t count,
                                  void* errmsg) {
    try {
        auto ix = PX(h);
        if (count != ix->count)
            throw std::runtime_error("assignment count != code count");
        for (uint64_t i = 0; i < count; ++i)
            if (assignments[i] < 0 || (uint32_t)assignments[i] >= ix->nlist)
                throw std::runtime_error("assignment out of range");
        ix->h_assign.assign(assignments, assignments + count);
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_set_assignments", e.what());
    }
}

void moann_ivf_pq_build(moann_ivf_pq_c h, void* errmsg) {
    try {
        auto ix = PX(h);
        std::lock_guard<std::mutex> lk(ix->mu);
        if (ix->built) return;
        if (ix->h_centroids.empty())
            throw std::runtime_error("no centroids");
        if (ix->h_codebooks.empty())
            throw std::runtime_error("no codebooks");
        if (ix->h_assign.size() != ix->count)
            throw std::runtime_error("no assignments");
        MOANN_HIP_CHECK(hipSetDevice(ix->device));

        /* CSR by list, stable order (same as IVF-Flat build) */
        std::vector<int64_t> loffsets(ix->nlist + 1, 0);
        for (auto a : ix->h_assign) loffsets[a + 1]++;
        for (uint32_t l = 0; l < ix->nlist; ++l) loffsets[l + 1] += loffsets[l];
        std::vector<int64_t> slot_rows(ix->count);
        {
            std::vector<int64_t> cur(loffsets.begin(), loffsets.end() - 1);
            for (uint64_t i = 0; i < ix->count; ++i)
                slot_rows[cur[ix->h_assign[i]]++] = (int64_t)i;
        }
        ix->list_slot_base.assign(loffsets.begin(), loffsets.end() - 1);
        ix->list_rows.resize(ix->nlist);
        ix->list_gbase.assign(ix->nlist + 1, 0);
        for (uint32_t l = 0; l < ix->nlist; ++l) {
            ix->list_rows[l] = (int32_t)(loffsets[l + 1] - loffsets[l]);
            ix->list_gbase[l + 1] =
                ix->list_gbase[l] + (ix->list_rows[l] + 63) / 64;
        }
        ix->total_groups = ix->list_gbase[ix->nlist];

        /* pack codes on device */
        std::vector<int64_t> group_rowbase(ix->total_groups);
        std::vector<int32_t> group_valid(ix->total_groups);
        for (uint32_t l = 0; l < ix->nlist; ++l)
            for (int64_t g = ix->list_gbase[l]; g < ix->list_gbase[l + 1];
                 ++g) {
                const int64_t gi = g - ix->list_gbase[l];
                group_rowbase[g] = loffsets[l] + gi * 64;
                group_valid[g] = (int32_t)std::min<int64_t>(
                    64, ix->list_rows[l] - gi * 64);
            }
        MOANN_HIP_CHECK(hipMalloc(&ix->d_codes, std::max<size_t>(
            1, (size_t)ix->total_groups * 64 * ix->nsub)));
        {
            MoannDevBuf d_src, d_rb, d_gv, d_sr;
            d_src.ensure(std::max<size_t>(1, ix->h_codes.size()));
            if (!ix->h_codes.empty())
                MOANN_HIP_CHECK(hipMemcpy(d_src.ptr, ix->h_codes.data(),
                                          ix->h_codes.size(),
                                          hipMemcpyHostToDevice));
            d_rb.ensure(std::max<size_t>(8, ix->total_groups * 8));
            d_gv.ensure(std::max<size_t>(4, ix->total_groups * 4));
            d_sr.ensure(std::max<size_t>(8, slot_rows.size() * 8));
            MOANN_HIP_CHECK(hipMemcpy(d_rb.ptr, group_rowbase.data(),
                                      ix->total_groups * 8,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMemcpy(d_gv.ptr, group_valid.data(),
                                      ix->total_groups * 4,
                                      hipMemcpyHostToDevice));
            if (!slot_rows.empty())
                MOANN_HIP_CHECK(hipMemcpy(d_sr.ptr, slot_rows.data(),
                                          slot_rows.size() * 8,
                                          hipMemcpyHostToDevice));
            if ((ix->nsub & 15) == 0)
                /* 16-byte units: the scan streams codes with dwordx4 */
                launch_bytes_pack16(d_src.as<uint8_t>(), (int)ix->nsub,
                                    d_rb.as<int64_t>(), d_gv.as<int32_t>(),
                                    d_sr.as<int64_t>(), ix->total_groups,
                                    ix->d_codes, ix->stream);
            else
                launch_pq_pack(d_src.as<uint8_t>(), (int)ix->nsub,
                               d_rb.as<int64_t>(), d_gv.as<int32_t>(),
                               d_sr.as<int64_t>(), ix->total_groups,
                               ix->d_codes, ix->stream);
            MOANN_HIP_CHECK(hipStreamSynchronize(ix->stream));
            MOANN_KCHECK("pq-pack");
        }

        /* centroids: row-major (LUT residuals) + interleaved (rank scan) */
        MOANN_HIP_CHECK(hipMalloc(&ix->d_cents_rows,
                                  (size_t)ix->nlist * ix->dpad * 4));
        if (ix->dim != ix->dpad)
            MOANN_HIP_CHECK(hipMemset(ix->d_cents_rows, 0,
                                      (size_t)ix->nlist * ix->dpad * 4));
        MOANN_HIP_CHECK(hipMemcpy2D(ix->d_cents_rows, (size_t)ix->dpad * 4,
                                    ix->h_centroids.data(),
                                    (size_t)ix->dim * 4, (size_t)ix->dim * 4,
                                    ix->nlist, hipMemcpyHostToDevice));
        {
            /* interleaved pack via the shared float pack kernel */
            MoannDevBuf d_rows2, d_rb, d_gv, d_sr;
            const int64_t cgroups = ((int64_t)ix->nlist + 63) / 64;
            std::vector<int64_t> rb(cgroups);
            std::vector<int32_t> gv(cgroups);
            std::vector<int64_t> sr(ix->nlist);
            std::iota(sr.begin(), sr.end(), 0);
            for (int64_t g = 0; g < cgroups; ++g) {
                rb[g] = g * 64;
                gv[g] = (int32_t)std::min<int64_t>(64,
                                                   (int64_t)ix->nlist - g * 64);
            }
            d_rows2.ensure((size_t)ix->nlist * ix->dim * 4);
            MOANN_HIP_CHECK(hipMemcpy(d_rows2.ptr, ix->h_centroids.data(),
                                      (size_t)ix->nlist * ix->dim * 4,
                                      hipMemcpyHostToDevice));
            d_rb.ensure(cgroups * 8);
            d_gv.ensure(cgroups * 4);
            d_sr.ensure(ix->nlist * 8);
            MOANN_HIP_CHECK(hipMemcpy(d_rb.ptr, rb.data(), cgroups * 8,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMemcpy(d_gv.ptr, gv.data(), cgroups * 4,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMemcpy(d_sr.ptr, sr.data(), ix->nlist * 8,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMalloc(&ix->d_cent_packed,
                                      (size_t)cgroups * 64 * ix->dpad * 4));
            launch_pack(d_rows2.as<float>(), (int)ix->dim, (int)ix->dpad,
                        d_rb.as<int64_t>(), d_gv.as<int32_t>(),
                        d_sr.as<int64_t>(), cgroups, ix->d_cent_packed,
                        ix->stream);
            MOANN_HIP_CHECK(hipStreamSynchronize(ix->stream));
            MOANN_KCHECK("pq-cent-pack");
            ix->cent_groups = cgroups;
        }

        /* codebooks + id/list maps */
        MOANN_HIP_CHECK(hipMalloc(&ix->d_codebooks,
                                  ix->h_codebooks.size() * 4));
        MOANN_HIP_CHECK(hipMemcpy(ix->d_codebooks, ix->h_codebooks.data(),
                                  ix->h_codebooks.size() * 4,
                                  hipMemcpyHostToDevice));
        /* per-list ADC B table: B_l[m][c] = ||c_l,m||^2 + 2 c_l,m.cb[m][c]
         * ([nlist][nsub*256] f32; 402 MB at nlist=4096, nsub=96) */
        MOANN_HIP_CHECK(hipMalloc(&ix->d_btab,
                                  (size_t)ix->nlist * ix->nsub * 256 * 4));
        launch_pq_tables(ix->d_cents_rows, nullptr, ix->d_codebooks,
                         (int)ix->nsub, (int)ix->sd, (int)ix->dpad,
                         (int64_t)ix->nlist, true, ix->d_btab, ix->stream);
        MOANN_HIP_CHECK(hipStreamSynchronize(ix->stream));
        MOANN_KCHECK("pq-btab");
        /* fold B into the per-row constant row_b[slot] = sum_m B_l[m][code],
         * then drop the B table — the search LUT is A_q only */
        {
            std::vector<int32_t> g2list(std::max<int64_t>(1, ix->total_groups));
            std::vector<int64_t> g2slot(std::max<int64_t>(1, ix->total_groups));
            std::vector<int32_t> g2valid(std::max<int64_t>(1, ix->total_groups));
            for (uint32_t l = 0; l < ix->nlist; ++l) {
                const int64_t lg = (ix->list_rows[l] + 63) / 64;
                for (int64_t gi = 0; gi < lg; ++gi) {
                    const int64_t g = ix->list_gbase[l] + gi;
                    g2list[g] = (int32_t)l;
                    g2slot[g] = ix->list_slot_base[l] + gi * 64;
                    g2valid[g] = (int32_t)std::min<int64_t>(
                        64, ix->list_rows[l] - gi * 64);
                }
            }
            MoannDevBuf d_l, d_s2, d_v;
            d_l.ensure(g2list.size() * 4);
            d_s2.ensure(g2slot.size() * 8);
            d_v.ensure(g2valid.size() * 4);
            MOANN_HIP_CHECK(hipMemcpy(d_l.ptr, g2list.data(),
                                      g2list.size() * 4,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMemcpy(d_s2.ptr, g2slot.data(),
                                      g2slot.size() * 8,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMemcpy(d_v.ptr, g2valid.data(),
                                      g2valid.size() * 4,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMalloc(&ix->d_rowb,
                                      std::max<uint64_t>(1, ix->count) * 4));
            launch_pq_rowb((ix->nsub & 15) == 0, ix->d_codes, ix->d_btab,
                           d_l.as<int32_t>(), d_s2.as<int64_t>(),
                           d_v.as<int32_t>(), ix->total_groups,
                           (int)ix->nsub, ix->d_rowb, ix->stream);
            MOANN_HIP_CHECK(hipStreamSynchronize(ix->stream));
            MOANN_KCHECK("pq-rowb");
            MOANN_HIP_CHECK(hipFree(ix->d_btab));
            ix->d_btab = nullptr;
        }
        /* |c_l|^2 + list geometry on device (per-query scan kernel) */
        {
            std::vector<float> cn(ix->nlist);
            for (uint32_t l = 0; l < ix->nlist; ++l) {
                double a = 0;
                const float* c = ix->h_centroids.data() + (size_t)l * ix->dim;
                for (uint32_t t = 0; t < ix->dim; ++t)
                    a += (double)c[t] * c[t];
                cn[l] = (float)a;
            }
            MOANN_HIP_CHECK(hipMalloc(&ix->d_cnorm2, ix->nlist * 4));
            MOANN_HIP_CHECK(hipMemcpy(ix->d_cnorm2, cn.data(), ix->nlist * 4,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMalloc(&ix->d_gbase_dev, ix->nlist * 8));
            MOANN_HIP_CHECK(hipMemcpy(ix->d_gbase_dev, ix->list_gbase.data(),
                                      ix->nlist * 8, hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMalloc(&ix->d_rows_dev, ix->nlist * 4));
            MOANN_HIP_CHECK(hipMemcpy(ix->d_rows_dev, ix->list_rows.data(),
                                      ix->nlist * 4, hipMemcpyHostToDevice));
        }
        std::vector<int64_t> id_by_slot(std::max<uint64_t>(1, ix->count));
        for (uint64_t sidx = 0; sidx < ix->count; ++sidx)
            id_by_slot[sidx] = ix->h_ids[slot_rows[sidx]];
        MOANN_HIP_CHECK(hipMalloc(&ix->d_id_by_slot, id_by_slot.size() * 8));
        MOANN_HIP_CHECK(hipMemcpy(ix->d_id_by_slot, id_by_slot.data(),
                                  id_by_slot.size() * 8,
                                  hipMemcpyHostToDevice));
        MOANN_HIP_CHECK(hipMalloc(&ix->d_list_slot_base, ix->nlist * 8));
        MOANN_HIP_CHECK(hipMemcpy(ix->d_list_slot_base,
                                  ix->list_slot_base.data(), ix->nlist * 8,
                                  hipMemcpyHostToDevice));
        ix->h_codes.clear();
        ix->h_codes.shrink_to_fit();
        ix->h_id_by_slot = std::move(id_by_slot);
        ix->built = true;
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_build", e.what());
    }
}

gpu_ivf_flat_search_res_t moann_ivf_pq_search(moann_ivf_pq_c h,
                                              const void* queries_data,
                                              uint64_t num_queries,
                                              uint32_t query_dimension,
                                              uint32_t limit,
                                              ivf_flat_search_params_t sp,
                                              void* errmsg) {
    gpu_ivf_flat_search_res_t out {nullptr};
    try {
        auto ix = PX(h);
        if (query_dimension != ix->dim)
            throw std::runtime_error("query dimension mismatch");
        out.result_ptr = pq_search(ix, (const float*)queries_data, false,
                                   num_queries, limit, sp.n_probes)
                             .release();
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_search", e.what());
    }
    return out;
}

gpu_ivf_flat_search_res_t moann_ivf_pq_search_filtered(
    moann_ivf_pq_c h, const void* queries, uint64_t num_queries,
    uint32_t query_dimension, uint32_t limit,
    ivf_flat_search_params_t sp, const uint32_t* filter_bitset,
    uint64_t bitset_words, void* errmsg) {
    gpu_ivf_flat_search_res_t out {nullptr};
    try {
        auto ix = PX(h);
        if (query_dimension != ix->dim)
            throw std::runtime_error("query dimension mismatch");
        if (bitset_words < (ix->count + 31) / 32)
            throw std::runtime_error("filter bitset too small");
        out.result_ptr = pq_search(ix, (const float*)queries, false,
                                   num_queries, limit, sp.n_probes,
                                   filter_bitset)
                             .release();
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_search_filtered", e.what());
    }
    return out;
}

void moann_ivf_pq_perf(moann_ivf_pq_c h, moann_perf_t* out) {
    *out = PX(h)->perf;
}

/* PQ persistence — same directory-manifest contract as the IVF index
 * (gpu_ivf_flat_save_dir/load_dir; index_base.hpp:239-249 precedent).
 * Blobs are SLOT-ordered; load replays them through the normal ingestion
 * (add_codes + set_assignments + build), which reproduces the identical
 * slot layout (stable sort by list). */
void moann_ivf_pq_save_dir(moann_ivf_pq_c h, const char* dir, void* errmsg) {
    try {
        auto ix = PX(h);
        std::lock_guard<std::mutex> lk(ix->mu);
        if (!ix->built) throw std::runtime_error("index not built");
        MOANN_HIP_CHECK(hipSetDevice(ix->device));
        std::string d(dir);
        auto wfile = [&](const std::string& name, const void* p, size_t n) {
            FILE* f = fopen((d + "/" + name).c_str(), "wb");
            if (!f) throw std::runtime_error("cannot write " + name);
            if (n && fwrite(p, 1, n, f) != n) {
                fclose(f);
                throw std::runtime_error("short write " + name);
            }
            fclose(f);
        };
        {
            char buf[256];
            snprintf(buf, sizeof buf,
                     "{\n  \"format\": \"moann-ivf-pq-v1\",\n"
                     "  \"dimension\": %u,\n  \"nsub\": %u,\n"
                     "  \"n_lists\": %u,\n  \"count\": %llu\n}\n",
                     ix->dim, ix->nsub, ix->nlist,
                     (unsigned long long)ix->count);
            wfile("manifest.json", buf, strlen(buf));
        }
        wfile("centroids.f32", ix->h_centroids.data(),
              ix->h_centroids.size() * 4);
        wfile("codebooks.f32", ix->h_codebooks.data(),
              ix->h_codebooks.size() * 4);
        wfile("ids.i64", ix->h_id_by_slot.data(), ix->count * 8);
        {
            std::vector<int32_t> a(ix->count);
            for (uint32_t l = 0; l < ix->nlist; ++l)
                for (int64_t s2 = ix->list_slot_base[l];
                     s2 < ix->list_slot_base[l] + ix->list_rows[l]; ++s2)
                    a[s2] = (int32_t)l;
            wfile("assign.i32", a.data(), ix->count * 4);
        }
        {
            /* codes.u8: unpack the interleaved image to slot-major rows */
            std::vector<int32_t> g2list;
            std::vector<int64_t> g2slot;
            std::vector<int32_t> g2valid;
            for (uint32_t l = 0; l < ix->nlist; ++l) {
                const int64_t lg = (ix->list_rows[l] + 63) / 64;
                for (int64_t gi = 0; gi < lg; ++gi) {
                    g2slot.push_back(ix->list_slot_base[l] + gi * 64);
                    g2valid.push_back((int32_t)std::min<int64_t>(
                        64, ix->list_rows[l] - gi * 64));
                }
            }
            MoannDevBuf d_gsb, d_gv, d_tmp;
            d_gsb.ensure(std::max<size_t>(8, g2slot.size() * 8));
            d_gv.ensure(std::max<size_t>(4, g2valid.size() * 4));
            MOANN_HIP_CHECK(hipMemcpy(d_gsb.ptr, g2slot.data(),
                                      g2slot.size() * 8,
                                      hipMemcpyHostToDevice));
            MOANN_HIP_CHECK(hipMemcpy(d_gv.ptr, g2valid.data(),
                                      g2valid.size() * 4,
                                      hipMemcpyHostToDevice));
            d_tmp.ensure(std::max<uint64_t>(1, ix->count) * ix->nsub);
            launch_unpack_bytes(ix->d_codes, (int)ix->nsub, (int)ix->nsub,
                                d_gsb.as<int64_t>(), d_gv.as<int32_t>(),
                                (int64_t)g2slot.size(), 0,
                                (int64_t)ix->count, d_tmp.as<uint8_t>(),
                                ix->stream);
            MOANN_HIP_CHECK(hipStreamSynchronize(ix->stream));
            std::vector<uint8_t> rows((size_t)ix->count * ix->nsub);
            MOANN_HIP_CHECK(hipMemcpy(rows.data(), d_tmp.ptr, rows.size(),
                                      hipMemcpyDeviceToHost));
            wfile("codes.u8", rows.data(), rows.size());
        }
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_save_dir", e.what());
    }
}

void moann_ivf_pq_load_dir(moann_ivf_pq_c h, const char* dir, void* errmsg) {
    try {
        auto ix = PX(h);
        if (ix->built) throw std::runtime_error("index already built");
        std::string d(dir);
        auto rfile = [&](const std::string& name, void* p, size_t n) {
            FILE* f = fopen((d + "/" + name).c_str(), "rb");
            if (!f) throw std::runtime_error("cannot read " + name);
            const size_t got = fread(p, 1, n, f);
            fclose(f);
            if (got != n) throw std::runtime_error("short read " + name);
        };
        char buf[512] = {0};
        rfile("manifest.json", buf, 0); /* existence check */
        {
            FILE* f = fopen((d + "/manifest.json").c_str(), "rb");
            size_t got = fread(buf, 1, sizeof buf - 1, f);
            fclose(f);
            (void)got;
        }
        unsigned dim = 0, nsub = 0, nlist = 0;
        unsigned long long count = 0;
        if (!strstr(buf, "moann-ivf-pq-v1"))
            throw std::runtime_error("unknown manifest format");
        sscanf(strstr(buf, "\"dimension\":"), "\"dimension\": %u", &dim);
        sscanf(strstr(buf, "\"nsub\":"), "\"nsub\": %u", &nsub);
        sscanf(strstr(buf, "\"n_lists\":"), "\"n_lists\": %u", &nlist);
        sscanf(strstr(buf, "\"count\":"), "\"count\": %llu", &count);
        if (dim != ix->dim || nsub != ix->nsub || nlist != ix->nlist)
            throw std::runtime_error(
                "manifest does not match the created index");
        if (count > ix->cap && ix->cap)
            throw std::runtime_error("manifest count exceeds capacity");
        std::vector<float> cents((size_t)nlist * dim);
        rfile("centroids.f32", cents.data(), cents.size() * 4);
        moann_ivf_pq_set_centroids(h, cents.data(), nlist, errmsg);
        std::vector<float> cb((size_t)nsub * 256 * (dim / nsub));
        rfile("codebooks.f32", cb.data(), cb.size() * 4);
        moann_ivf_pq_set_codebooks(h, cb.data(), errmsg);
        std::vector<int64_t> ids(count);
        rfile("ids.i64", ids.data(), count * 8);
        std::vector<uint8_t> codes((size_t)count * nsub);
        rfile("codes.u8", codes.data(), codes.size());
        moann_ivf_pq_add_codes(h, codes.data(), count, ids.data(), errmsg);
        std::vector<int32_t> assign(count);
        rfile("assign.i32", assign.data(), count * 4);
        moann_ivf_pq_set_assignments(h, assign.data(), count, errmsg);
        moann_ivf_pq_build(h, errmsg);
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_load_dir", e.what());
    }
}

void moann_ivf_pq_destroy(moann_ivf_pq_c h, void* errmsg) {
    try {
        delete PX(h);
    } catch (const std::exception& e) {
        moann_set_errmsg(errmsg, "moann_ivf_pq_destroy", e.what());
    }
}

}  /* extern "C" */
