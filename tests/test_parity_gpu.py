"""GPU parity tests: the HIP product path vs the oracle (CPU restatement of
the reference algorithms, pinned against usearch/SimSIMD) on the same seeded
inputs. Run on a real MI355X (`pytest -m gpu`).

Parity bar (BASELINE.json north star): neighbor ids bit-exact except at FP
ties (where distances agree within 1e-5 relative); distances within 1e-5
relative."""

import glob
import os

import numpy as np
import pytest

from oracle import oracle as orc

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")
RTOL = 1e-5


def _assert_parity(ids, dists, ref_ids, ref_dists, ctx=""):
    ids = np.asarray(ids)
    ref_ids = np.asarray(ref_ids)
    d = np.asarray(dists, dtype=np.float64)
    rd = np.asarray(ref_dists, dtype=np.float64)
    np.testing.assert_allclose(d, rd, rtol=RTOL, atol=1e-6, err_msg=ctx)
    mism = ids != ref_ids
    if mism.any():
        da, db = d[mism], rd[mism]
        rel = np.abs(da - db) / np.maximum(np.maximum(np.abs(da), np.abs(db)),
                                           1e-30)
        assert (rel < RTOL).all(), (
            f"{ctx}: {mism.sum()} id mismatches beyond FP ties "
            f"(worst rel dist err {rel.max():.2e})")


def _build_from_fixture(z, engine):
    ix = engine.IvfFlatIndex(z["vecs"].shape[1], z["centroids"].shape[0],
                             metric={0: "l2", 1: "l2sq", 2: "ip", 3: "cos",
                                     4: "l1"}[int(z["metric"])],
                             capacity=z["vecs"].shape[0])
    ix.add(z["vecs"], ids=z["ids"])
    ix.set_centroids(z["centroids"])
    ix.set_assignments(z["assign"].astype(np.int32))
    ix.build()
    return ix


def test_ivf_golden_fixtures():
    """Every committed IVF fixture: HIP results == oracle results."""
    from matrixone_amd import engine
    for path in sorted(glob.glob(os.path.join(GOLDEN, "ivf_*.npz"))):
        z = np.load(path)
        ix = _build_from_fixture(z, engine)
        ids, dists = ix.search(z["queries"], int(z["k"]), int(z["probe"]))
        _assert_parity(ids, dists, z["out_ids"], z["out_dists"], ctx=path)
        ix.close()


def test_bf_golden_fixtures():
    from matrixone_amd import engine
    for path in sorted(glob.glob(os.path.join(GOLDEN, "bf_*.npz"))):
        z = np.load(path)
        ids, dists = engine.brute_force_search(
            z["base"], z["queries"], int(z["k"]),
            metric={1: "l2sq", 2: "ip", 3: "cos"}[int(z["metric"])])
        _assert_parity(ids, dists, z["out_ids"], z["out_dists"], ctx=path)


@pytest.mark.parametrize("metric,orig_l2", [
    ("l2sq", False), ("l2", True), ("ip", False), ("cos", False),
    ("l1", False)])
def test_ivf_all_metrics_live(metric, orig_l2):
    """Config-1 shape (BASELINE.json configs[0], shrunk): 20k x 128,
    nlist=64, nprobe=8 — oracle runs in seconds; every metric."""
    from matrixone_amd import engine
    mmap = {"l2sq": orc.METRIC_L2SQ, "l2": orc.METRIC_L2,
            "ip": orc.METRIC_IP, "cos": orc.METRIC_COS, "l1": orc.METRIC_L1}
    rng = np.random.Generator(np.random.PCG64(100))
    n, d, nlist, nq, probe, k = 20000, 128, 64, 32, 8, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = np.empty(n, dtype=np.int64)
    for c0 in range(0, n, 4096):
        blk = vecs[c0:c0 + 4096].astype(np.float64)
        d2 = ((blk[:, None, :] - cents[None].astype(np.float64)) ** 2).sum(-1)
        assign[c0:c0 + 4096] = d2.argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) + 7
    queries = rng.standard_normal((nq, d), dtype=np.float32)

    idx = orc.IvfIndex(cents, vecs, assign, ids=ids64)
    ref_ids, ref_dists = idx.search(mmap[metric], queries, probe, k,
                                    orig_l2=orig_l2)
    ix = engine.IvfFlatIndex(d, nlist, metric=metric, capacity=n)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    _assert_parity(ids, dists, ref_ids, ref_dists, ctx=metric)
    ix.close()


def test_probe_set_matches_oracle():
    """The centroid ranking (list selection) itself must match the oracle —
    this is what makes candidate sets identical (ivfflat/search.go:265-308)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(200))
    n, d, nlist, nq, probe, k = 5000, 64, 32, 16, 6, 2000
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    idx = orc.IvfIndex(cents, vecs, assign)
    # k=2000 > all candidates in 6 lists is possible; both sides must agree
    # on the full candidate set and its padding.
    ref_ids, ref_dists = idx.search(orc.METRIC_L2SQ, queries, probe, k)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    # padded tails must agree exactly
    np.testing.assert_array_equal(ids == -1, ref_ids == -1)
    _assert_parity(ids, dists, ref_ids, ref_dists, ctx="probe-set")
    ix.close()


def test_edge_cases():
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(300))
    # empty lists, k > n, single query, dim 7 (non-multiple of 4)
    n, d, nlist = 100, 7, 8
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = rng.standard_normal((nlist, d), dtype=np.float32)
    assign = np.zeros(n, dtype=np.int64)  # everything in list 0 — 7 empty
    q = rng.standard_normal((1, d), dtype=np.float32)
    idx = orc.IvfIndex(cents, vecs, assign)
    ref_ids, ref_dists = idx.search(orc.METRIC_L2SQ, q, 8, 200)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(q, 200, 8)
    np.testing.assert_array_equal(ids == -1, ref_ids == -1)
    _assert_parity(ids, dists, ref_ids, ref_dists, ctx="edge")
    ix.close()


def test_duplicate_vectors_ties():
    """Exact duplicates: distance values must agree exactly; the id SET at
    the tie boundary must be a valid top-k set."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(400))
    base = rng.standard_normal((64, 16), dtype=np.float32)
    vecs = np.repeat(base, 8, axis=0)  # 512 rows, every vector x8
    cents = base[:4].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    q = vecs[:3] + 1e-3
    idx = orc.IvfIndex(cents, vecs, assign)
    ref_ids, ref_dists = idx.search(orc.METRIC_L2SQ, q, 4, 20)
    ix = engine.IvfFlatIndex(16, 4, metric="l2sq", capacity=len(vecs))
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(q, 20, 4)
    # accumulation order differs (oracle 8-wide pairwise vs GPU float4
    # sequential) so distances agree to 1e-5 relative, not bitwise; the tie
    # GROUPS must still line up position by position.
    np.testing.assert_allclose(dists, ref_dists, rtol=1e-5, atol=1e-6)
    for row_g, row_o in zip(dists, ref_dists):
        np.testing.assert_array_equal(np.diff(row_g) == 0, np.diff(row_o) == 0)
    ix.close()


# (HNSW GPU coverage lives in tests/test_hnsw_gpu.py)


def test_filtered_search():
    """Membership-filtered search (§8a4): results equal the oracle's full
    candidate ranking post-filtered to the allowed id set."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(500))
    n, d, nlist, nq, probe, k = 8000, 48, 16, 12, 4, 15
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) * 3 + 11
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    allowed = ids64[rng.random(n) < 0.2]

    # oracle expectation: rank ALL candidates (k = n), keep allowed, truncate
    idx = orc.IvfIndex(cents, vecs, assign, ids=ids64)
    all_ids, all_d = idx.search(orc.METRIC_L2SQ, queries, probe, 4000)
    aset = set(allowed.tolist())
    exp_ids = np.full((nq, k), -1, dtype=np.int64)
    exp_d = np.full((nq, k), np.finfo(np.float32).max, dtype=np.float32)
    for i in range(nq):
        keep = [(all_d[i, j], all_ids[i, j]) for j in range(all_ids.shape[1])
                if all_ids[i, j] >= 0 and all_ids[i, j] in aset]
        for j, (dd, ii) in enumerate(keep[:k]):
            exp_ids[i, j] = ii
            exp_d[i, j] = dd

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    words = ix.filter_bitset(allowed)
    gids, gdists = ix.search_filtered(queries, k, probe, words)
    np.testing.assert_array_equal(gids == -1, exp_ids == -1)
    _assert_parity(gids, gdists, exp_ids, exp_d, ctx="filtered")
    # unfiltered still works after a filtered call
    u_ids, _ = ix.search(queries, k, probe)
    assert (u_ids >= 0).all()
    ix.close()


def test_save_load_dir(tmp_path):
    """save_dir -> fresh handle + load_dir reproduces identical results
    (the gpu_ivf_flat_save_dir/load_dir persistence contract)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(600))
    n, d, nlist, nq, probe, k = 5000, 40, 16, 8, 4, 12
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) + 5
    queries = rng.standard_normal((nq, d), dtype=np.float32)

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids1, d1 = ix.search(queries, k, probe)
    ix.save_dir(str(tmp_path))
    ix.close()

    ix2 = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix2.load_dir(str(tmp_path))
    ids2, d2 = ix2.search(queries, k, probe)
    np.testing.assert_array_equal(ids1, ids2)
    np.testing.assert_array_equal(d1, d2)
    # mismatched geometry must be rejected
    ix3 = engine.IvfFlatIndex(d, nlist * 2, metric="l2sq", capacity=n)
    with pytest.raises(engine.MoannError):
        ix3.load_dir(str(tmp_path))
    ix2.close()
    ix3.close()


def test_dynamic_batching():
    """search_async calls inside the batch window fuse into one device
    search and return per-request results identical to the unbatched path
    (dynamic_batching.hpp:19-60 contract)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(700))
    n, d, nlist = 20000, 64, 32
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[:nlist].copy()
    assign = ((vecs[:, None, :] - cents[None]) ** 2).sum(-1).argmin(1)
    ix = engine.IvfFlatIndex(d, nlist, capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    qs = rng.standard_normal((16, d), dtype=np.float32)
    ref, ref_d = ix.search(qs, 5, 4)
    ix.set_batch_window(2000)
    jobs = [ix.search_async(qs[i:i + 1], 5, 4) for i in range(16)]
    for i, j in enumerate(jobs):
        ix._async_shape = (1, 5)
        ids, dists = ix.search_wait(j)
        np.testing.assert_array_equal(ids[0], ref[i])
        np.testing.assert_array_equal(dists[0], ref_d[i])
    ix.set_batch_window(0)
    ids2, _ = ix.search(qs, 5, 4)
    np.testing.assert_array_equal(ids2, ref)
    ix.close()


def test_pairwise_distance():
    """1xN pairwise batch (the SQL builtin shape, func_binary.go:9127) vs
    the oracle distance kernels, all metrics."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(800))
    for metric, om in [("l2sq", orc.METRIC_L2SQ), ("l2", orc.METRIC_L2),
                       ("ip", orc.METRIC_IP), ("cos", orc.METRIC_COS),
                       ("l1", orc.METRIC_L1)]:
        rows = rng.standard_normal((500, 131), dtype=np.float32)
        q = rng.standard_normal(131, dtype=np.float32)
        got = engine.pairwise_distance(metric, q, rows)
        want = np.array([orc.distance(om, rows[i], q) for i in range(500)])
        np.testing.assert_allclose(got, want, rtol=2e-5, atol=2e-5,
                                   err_msg=metric)


@pytest.mark.parametrize("qtype,metric", [
    ("int8", "l2"), ("int8", "ip"), ("uint8", "l2sq"), ("int8", "cos"),
    ("int8", "l1")])
def test_quantized_storage(qtype, metric):
    """INT8/UINT8 storage (the reference's wired btype/qtype combos): entries
    and query quantized by the same affine map, EXACT integer distances in
    the quantized domain, raw/mul^2 rescale + sqrt transform — vs the oracle
    quantized restatement."""
    from matrixone_amd import engine
    mmap = {"l2sq": orc.METRIC_L2SQ, "l2": orc.METRIC_L2,
            "ip": orc.METRIC_IP, "cos": orc.METRIC_COS, "l1": orc.METRIC_L1}
    rng = np.random.Generator(np.random.PCG64(900))
    n, d, nlist, nq, probe, k = 12000, 96, 32, 16, 6, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    vmin, vmax = np.percentile(vecs, 0.1), np.percentile(vecs, 99.9)
    uns = qtype == "uint8"
    mul, add = (orc.uint8_params(vmin, vmax) if uns
                else orc.int8_params(vmin, vmax))

    idx = orc.IvfIndex(cents, vecs, assign)
    ref_ids, ref_d = orc.ivf_search_quantized(
        idx, mmap[metric], queries, probe, k, mul, add, unsigned=uns,
        orig_l2=(metric == "l2"))

    ix = engine.IvfFlatIndex(d, nlist, metric=metric, capacity=n, qtype=qtype)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.set_quantizer(vmin, vmax)
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    _assert_parity(ids, dists, ref_ids, ref_d, ctx=f"{qtype}/{metric}")
    ix.close()


def test_quantized_storage_padded_dim():
    """dim not divisible by 4: padded columns must stay identity under
    quantization (q(0) != 0 would poison IP/cos sums)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(910))
    n, d, nlist, nq, probe, k = 4000, 67, 16, 8, 4, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    vmin, vmax = float(vecs.min()), float(vecs.max())
    mul, add = orc.int8_params(vmin, vmax)
    idx = orc.IvfIndex(cents, vecs, assign)
    ref_ids, ref_d = orc.ivf_search_quantized(
        idx, orc.METRIC_IP, queries, probe, k, mul, add)
    ix = engine.IvfFlatIndex(d, nlist, metric="ip", capacity=n, qtype="int8")
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.set_quantizer(vmin, vmax)
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    _assert_parity(ids, dists, ref_ids, ref_d, ctx="int8-ip-d67")
    ix.close()


def test_asm768_scan_parity():
    """The hand-scheduled d=768 L2 asm kernel (scan_asm768.hip, the default
    path for the flagship shape): results == oracle on ragged lists (rows not
    a multiple of 64), >16 queries (multiple tiles incl. a short tail tile),
    orig-l2 transform, and a filtered pass. The centroid rank scan takes the
    same kernel, so probe sets are exercised too."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(768))
    n, d, nlist, nq, probe, k = 30000, 768, 24, 37, 6, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = np.empty(n, dtype=np.int64)
    for c0 in range(0, n, 2048):
        blk = vecs[c0:c0 + 2048].astype(np.float64)
        d2 = ((blk[:, None, :] - cents[None].astype(np.float64)) ** 2).sum(-1)
        assign[c0:c0 + 2048] = d2.argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) + 3
    queries = rng.standard_normal((nq, d), dtype=np.float32)

    idx = orc.IvfIndex(cents, vecs, assign, ids=ids64)
    ref_ids, ref_d = idx.search(orc.METRIC_L2, queries, probe, k, orig_l2=True)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2", capacity=n)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    _assert_parity(ids, dists, ref_ids, ref_d, ctx="asm768")

    # filtered: FLT_MAX substitution happens inside the asm kernel's
    # epilogue. Expectation = oracle's full ranking post-filtered.
    allowed = ids64[rng.random(n) < 0.25]
    all_ids, all_d = idx.search(orc.METRIC_L2, queries, probe, 2000,
                                orig_l2=True)
    aset = set(allowed.tolist())
    exp_ids = np.full((nq, k), -1, dtype=np.int64)
    exp_d = np.full((nq, k), np.finfo(np.float32).max, dtype=np.float32)
    for i in range(nq):
        keep = [(all_d[i, j], all_ids[i, j]) for j in range(all_ids.shape[1])
                if all_ids[i, j] >= 0 and all_ids[i, j] in aset]
        for j, (dd, ii) in enumerate(keep[:k]):
            exp_ids[i, j] = ii
            exp_d[i, j] = dd
    bits = ix.filter_bitset(allowed)
    fids, fdists = ix.search_filtered(queries, k, probe, bits)
    _assert_parity(fids, fdists, exp_ids, exp_d, ctx="asm768-filtered")
    ix.close()


def test_pipelined_submit_collect():
    """moann_ivf_flat_search_submit/result_wait (depth-2 overlap): results
    of interleaved pipelined batches == the synchronous search path on the
    same inputs, and FIFO order is preserved."""
    import torch
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(41))
    n, d, nlist, probe, k = 20000, 768, 32, 6, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = np.empty(n, dtype=np.int64)
    for c0 in range(0, n, 4096):
        blk = vecs[c0:c0 + 4096].astype(np.float64)
        d2 = ((blk[:, None, :] - cents[None].astype(np.float64)) ** 2).sum(-1)
        assign[c0:c0 + 4096] = d2.argmin(1)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    batches = [torch.from_numpy(
        rng.standard_normal((64, d), dtype=np.float32)).cuda()
        for _ in range(5)]
    seq = [ix.search_device(q, k, probe) for q in batches]
    # depth-2 pipeline, FIFO collect
    tickets = []
    got = []
    for i, q in enumerate(batches):
        tickets.append(ix.search_submit(q, k, probe))
        if len(tickets) == 2:
            got.append(ix.search_collect(tickets.pop(0)))
    while tickets:
        got.append(ix.search_collect(tickets.pop(0)))
    for i in range(5):
        np.testing.assert_array_equal(got[i][0], seq[i][0], err_msg=f"b{i}")
        np.testing.assert_allclose(got[i][1], seq[i][1], rtol=1e-6)
    # a second round still works (ctx reuse path)
    ids2, _ = ix.search_collect(ix.search_submit(batches[0], k, probe))
    np.testing.assert_array_equal(ids2, seq[0][0])
    ix.close()


def test_docfiltered_search_gpu():
    """End-to-end docfilter-gated search (§8f3): a serialized sorted64 /
    croaring / cbitmap filter drives IVF search_docfiltered (and the HNSW
    variant) to exactly the results of the equivalent explicit bitset."""
    from matrixone_amd import engine
    from oracle import docfilter as DF
    rng = np.random.Generator(np.random.PCG64(64))
    n, d, nlist, nq, probe, k = 9000, 64, 16, 10, 5, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) * 5 + 17
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    allowed = ids64[rng.random(n) < 0.3]

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    exp_words = ix.filter_bitset(allowed)
    exp_ids, exp_d = ix.search_filtered(queries, k, probe, exp_words)
    for kind, payload in [
        (engine.FILTER_SORTED64, DF.build_sorted64(allowed)),
        (engine.FILTER_CROARING, DF.build_croaring(allowed)),
        (engine.FILTER_CBITMAP, DF.build_cbitmap(allowed)),
    ]:
        gids, gdists = ix.search_docfiltered(queries, k, probe, kind, payload)
        np.testing.assert_array_equal(gids, exp_ids)
        np.testing.assert_allclose(gdists, exp_d)
    # bloom: approximate — supersets allowed, never false negatives
    bpay = DF.build_bloom(allowed, nbits=1 << 16, k=7, seed=11)
    bids, _ = ix.search_docfiltered(queries, k, probe,
                                    engine.FILTER_BLOOM, bpay)
    aset = set(int(x) for x in allowed)
    hit = [int(x) in aset for x in exp_ids[exp_ids >= 0]]
    assert all(hit)
    ix.close()


def test_save_load_dir_quantized(tmp_path):
    """Narrow-storage persistence: save writes the QUANTIZED bytes + the
    quantizer in the manifest; load dequantizes and re-quantizes with the
    saved parameters — an exact round trip (identical search results)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(55))
    n, d, nlist, nq, probe, k = 8000, 96, 16, 10, 5, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) + 13
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    vmin, vmax = float(vecs.min()), float(vecs.max())

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                             qtype="int8")
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.set_quantizer(vmin, vmax)
    ix.build()
    ref_ids, ref_d = ix.search(queries, k, probe)
    ix.save_dir(str(tmp_path))
    ix.close()

    ix2 = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                              qtype="int8")
    ix2.load_dir(str(tmp_path))
    ids2, d2 = ix2.search(queries, k, probe)
    np.testing.assert_array_equal(ids2, ref_ids)
    np.testing.assert_allclose(d2, ref_d)
    # mismatched storage type fails loudly
    ix3 = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    with pytest.raises(engine.MoannError):
        ix3.load_dir(str(tmp_path))
    ix2.close()
    ix3.close()


def test_refine_exactness():
    """Two-stage scan (enable_refine): with R >= every candidate count the
    result must EQUAL the exact one-pass search (every candidate re-ranked
    with exact f32), and at small R the returned distances are still the
    EXACT f32 distances of the returned ids."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(21))
    n, d, nlist, nq, probe, k = 20000, 768, 64, 16, 4, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = np.empty(n, dtype=np.int64)
    for c0 in range(0, n, 4096):
        blk = vecs[c0:c0 + 4096].astype(np.float64)
        d2 = ((blk[:, None, :] - cents[None].astype(np.float64)) ** 2).sum(-1)
        assign[c0:c0 + 4096] = d2.argmin(1)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()

    # R = 4096 > max candidates per query (~20000*4/64 ~ 1250): identical.
    # The refine search runs FIRST on the fresh index: a prior exact search
    # over the same queries leaves correct f32 distances in the reused
    # candidate buffer, which masked the round-1 QT-16/QT-8 job-geometry
    # bug (unwritten tile lanes read stale-but-correct values).
    ix.enable_refine(4096)
    rid, rd = ix.search(queries, k, probe)
    ix.enable_refine(0)
    exact_ids, exact_d = ix.search(queries, k, probe)
    np.testing.assert_array_equal(rid, exact_ids)
    np.testing.assert_allclose(rd, exact_d, rtol=2e-5, atol=2e-5)

    # small R: whatever ids come back carry their EXACT f32 distances
    ix.enable_refine(32)
    sid, sd = ix.search(queries, k, probe)
    ix.enable_refine(0)  # exact reference for distance lookup
    ref_all, ref_alld = ix.search(queries, 1200, probe)
    for i in range(nq):
        lut = {int(a): float(b) for a, b in zip(ref_all[i], ref_alld[i])
               if a >= 0}
        for j in range(k):
            if sid[i, j] < 0:
                continue
            assert int(sid[i, j]) in lut
            np.testing.assert_allclose(sd[i, j], lut[int(sid[i, j])],
                                       rtol=2e-5, atol=2e-5)
    ix.close()


def test_search_with_tail():
    """CDC-tail overflow merge (§8f4, ivfpq/search_gpu.go:222-428 shape):
    index search + brute force over the unindexed tail, merged == oracle
    over the UNION corpus at full probe."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(31))
    n, ntail, d, nlist, nq, k = 6000, 500, 48, 8, 10, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    tail = rng.standard_normal((ntail, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids = np.arange(n, dtype=np.int64)
    tail_ids = np.arange(n, n + ntail, dtype=np.int64)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs, ids=ids)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    got_i, got_d = engine.search_with_tail(ix, queries, k, nlist, tail,
                                           tail_ids)
    union = np.concatenate([vecs, tail])
    union_ids = np.concatenate([ids, tail_ids])
    exp_i, exp_d = orc.bruteforce(orc.METRIC_L2SQ, union, queries, k,
                                  ids=union_ids)
    _assert_parity(got_i, got_d, exp_i, exp_d, ctx="tail-merge")
    # empty tail degenerates to the plain index search
    a_i, a_d = engine.search_with_tail(ix, queries, k, nlist, None, None)
    b_i, b_d = ix.search(queries, k, nlist)
    np.testing.assert_array_equal(a_i, b_i)
    ix.close()


@pytest.mark.parametrize("qtype,metric", [
    ("f16", "l2sq"), ("f16", "cos"), ("f16", "ip"), ("f16", "l1"),
    ("bf16", "l2sq"), ("bf16", "ip"), ("bf16", "cos")])
def test_half_storage(qtype, metric):
    """f16/bf16 plain-cast storage (quantizer.go:50-58): the HIP half scan
    == the oracle run over RTNE-decoded entries and query
    (relation_search.go:56-63, distance_func_narrow.go:27-52)."""
    from matrixone_amd import engine
    mmap = {"l2sq": orc.METRIC_L2SQ, "ip": orc.METRIC_IP,
            "cos": orc.METRIC_COS, "l1": orc.METRIC_L1}
    rng = np.random.Generator(np.random.PCG64(1600 + len(qtype)))
    n, d, nlist, nq, probe, k = 12000, 96, 32, 12, 6, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) + 5
    queries = rng.standard_normal((nq, d), dtype=np.float32)

    idx = orc.IvfIndex(cents, vecs, assign, ids=ids64)
    ref_ids, ref_d = orc.ivf_search_half(idx, mmap[metric], queries, probe,
                                         k, bf16=(qtype == "bf16"))
    ix = engine.IvfFlatIndex(d, nlist, metric=metric, capacity=n,
                             qtype=qtype)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    _assert_parity(ids, dists, ref_ids, ref_d, ctx=f"{qtype}/{metric}")
    ix.close()


def test_half_storage_odd_dim_and_saveload(tmp_path):
    """dim not a multiple of 8 (dpad rounds to 8 for halves) + persistence
    round trip for f16 storage."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(77))
    n, d, nlist, nq, probe, k = 5000, 67, 16, 8, 4, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    idx = orc.IvfIndex(cents, vecs, assign)
    ref_ids, ref_d = orc.ivf_search_half(idx, orc.METRIC_L2SQ, queries,
                                         probe, k, bf16=False)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                             qtype="f16")
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, k, probe)
    _assert_parity(ids, dists, ref_ids, ref_d, ctx="f16-d67")
    ix.save_dir(str(tmp_path))
    ix2 = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                              qtype="f16")
    ix2.load_dir(str(tmp_path))
    ids2, d2 = ix2.search(queries, k, probe)
    np.testing.assert_array_equal(ids2, ids)
    np.testing.assert_allclose(d2, dists)
    ix.close()
    ix2.close()


def test_pred_filtered_search():
    """SQL residual-predicate gated search (§8f3): preds -> slot bitset
    (ADD-order columns permuted to slots) == the same search with an
    explicitly built id bitset."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(91))
    n, d, nlist, nq, probe, k = 8000, 48, 16, 10, 5, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    ids64 = rng.permutation(n).astype(np.int64) + 11
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    price = rng.integers(0, 1000, n).astype(np.int64)  # ADD-order column

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs, ids=ids64)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    preds = '[{"col":0,"op":"between","lo":100,"hi":400}]'
    gids, gdists = ix.search_pred_filtered(
        queries, k, probe, [("price", engine.COL_I64)], [price], preds)
    allowed = ids64[(price >= 100) & (price <= 400)]
    eids, edists = ix.search_filtered(queries, k, probe,
                                     ix.filter_bitset(allowed))
    np.testing.assert_array_equal(gids, eids)
    np.testing.assert_allclose(gdists, edists)
    # PK predicate composes with the column predicate
    sel = set(int(x) for x in allowed[:50])
    preds2 = ('[{"col":0,"op":">=","val":100},'
              '{"col":0,"op":"<=","val":400},'
              '{"col":-1,"op":"in","vals":[%s]}]'
              % ",".join(str(x) for x in sorted(sel)))
    g2, d2 = ix.search_pred_filtered(
        queries, k, probe, [("price", engine.COL_I64)], [price], preds2)
    e2, ed2 = ix.search_filtered(queries, k, probe,
                                 ix.filter_bitset(np.array(sorted(sel))))
    np.testing.assert_array_equal(g2, e2)
    ix.close()


def test_ip_sign_at_c_abi_matches_reference_contract():
    """The raw C ABI must emit the TRUE inner product +a.b for IP results
    (the reference flips cuvs' negated IP before it crosses the C ABI —
    cgo/cuvs/index_base.hpp:355-375 transform_distance), while the host
    mirror restores MO's -a.b convention (distance_func.go:174-207). So
    raw-ctypes distances == -(engine-level distances) except sentinels."""
    import ctypes as ct
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(4242))
    n, d, nlist, nq, probe, k = 5000, 96, 32, 8, 4, 12
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    ix = engine.IvfFlatIndex(d, nlist, metric="ip", capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    eng_ids, eng_dists = ix.search(queries, k, probe)

    L = engine.lib()
    err = engine._Err()
    res = L.gpu_ivf_flat_search(ix._h, queries.ctypes.data, nq, d, k,
                                engine.SearchParams(probe), err.ref)
    err.check("raw search")
    raw_ids = np.empty((nq, k), dtype=np.int64)
    raw_dists = np.empty((nq, k), dtype=np.float32)
    L.gpu_ivf_flat_get_neighbors(res.result_ptr, nq * k, raw_ids.ctypes.data)
    L.gpu_ivf_flat_get_distances(res.result_ptr, nq * k, raw_dists.ctypes.data)
    L.gpu_ivf_flat_free_result(res.result_ptr)
    np.testing.assert_array_equal(raw_ids, eng_ids)
    sent = np.float32(np.finfo(np.float32).max)
    live = (raw_dists != sent) & (raw_dists != -sent)
    np.testing.assert_array_equal(live, (eng_dists != sent) & (eng_dists != -sent))
    np.testing.assert_array_equal(raw_dists[live], -eng_dists[live])
    # +a.b really is the inner product of the returned neighbor
    q0n = int(raw_ids[0, 0])  # default ids are 0..n-1 (row order)
    want = float(np.dot(queries[0].astype(np.float64), vecs[q0n].astype(np.float64)))
    assert abs(raw_dists[0, 0] - want) <= 1e-4 * max(1.0, abs(want))
    ix.close()
