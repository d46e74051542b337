"""IVF-PQ tests: CPU (oracle self-checks) + GPU parity vs the oracle ADC
restatement on identical codebooks/codes."""

import numpy as np
import pytest

from oracle import oracle as orc
from oracle import pq as opq


def _make(n=6000, d=32, nsub=8, nlist=16, seed=31):
    rng = np.random.Generator(np.random.PCG64(seed))
    centers = rng.standard_normal((60, d), dtype=np.float32)
    comp = rng.integers(0, 60, n)
    vecs = (centers[comp] + 0.5 * rng.standard_normal((n, d))).astype(np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    cb = opq.train_pq(vecs, cents, assign, nsub, iters=5, seed=7)
    codes = opq.encode_pq(vecs, cents, assign, cb)
    queries = (centers[rng.integers(0, 60, 16)] +
               0.5 * rng.standard_normal((16, d))).astype(np.float32)
    return vecs, cents, assign, cb, codes, queries


def test_pq_oracle_recall_reasonable():
    """ADC over trained codebooks must approximate exact search on clustered
    data (the recall-harness shape of pkg/cuvs/recall_test.go:38-75)."""
    vecs, cents, assign, cb, codes, queries = _make()
    idx = opq.PqIndex(cents, cb, codes, assign)
    gt, _ = orc.bruteforce(orc.METRIC_L2SQ, vecs, queries, 10)

    def recall(probe):
        ids, _ = idx.search(queries, probe=probe, k=10)
        return np.mean([len(set(ids[i]).intersection(gt[i])) / 10
                        for i in range(len(queries))])

    r_full = recall(16)  # every list probed: loss is quantization only
    assert r_full > 0.4, r_full
    # probing more lists can only help (monotone in probe)
    assert recall(8) <= r_full + 1e-9


def test_pq_encode_roundtrip_error_bounded():
    vecs, cents, assign, cb, codes, _ = _make(n=2000)
    nsub, _, sd = cb.shape
    recon = cents[assign].copy()
    for m in range(nsub):
        recon[:, m * sd:(m + 1) * sd] += cb[m][codes[:, m]]
    err = np.linalg.norm(recon - vecs, axis=1)
    base = np.linalg.norm(vecs - cents[assign], axis=1)
    assert err.mean() < base.mean()  # codes explain residual variance


@pytest.mark.gpu
def test_pq_gpu_vs_oracle():
    from matrixone_amd import engine
    vecs, cents, assign, cb, codes, queries = _make()
    idx = opq.PqIndex(cents, cb, codes, assign)
    ref_ids, ref_d = idx.search(queries, probe=6, k=12)

    ix = engine.IvfPqIndex(vecs.shape[1], cb.shape[0], cents.shape[0],
                           capacity=len(vecs))
    ix.set_centroids(cents)
    ix.set_codebooks(cb)
    ix.add_codes(codes)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, 12, 6)
    np.testing.assert_allclose(dists, ref_d, rtol=2e-5, atol=2e-5)
    mism = ids != ref_ids
    if mism.any():
        rel = np.abs(dists[mism].astype(np.float64) - ref_d[mism]) / \
            np.maximum(np.abs(ref_d[mism]), 1e-30)
        assert (rel < 2e-5).all()
    ix.close()


@pytest.mark.gpu
def test_pq_gpu_custom_ids_and_edges():
    from matrixone_amd import engine
    vecs, cents, assign, cb, codes, queries = _make(n=500, nlist=8)
    ids64 = np.arange(500, dtype=np.int64) * 2 + 3
    idx = opq.PqIndex(cents[:8], cb, codes, assign, ids=ids64)
    ref_ids, ref_d = idx.search(queries, probe=8, k=600)  # k > n

    ix = engine.IvfPqIndex(vecs.shape[1], cb.shape[0], 8, capacity=500)
    ix.set_centroids(cents[:8])
    ix.set_codebooks(cb)
    ix.add_codes(codes, ids=ids64)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    ids, dists = ix.search(queries, 600, 8)
    np.testing.assert_array_equal(ids == -1, ref_ids == -1)
    np.testing.assert_allclose(dists, ref_d, rtol=2e-5, atol=2e-5)
    ix.close()
