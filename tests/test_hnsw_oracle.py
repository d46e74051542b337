"""CPU tests: the usearch shim, the graph export, and the numpy restatement
of usearch's HNSW search must agree with usearch's own results on the same
index (oracle/hnsw.py module docstring for algorithm cites)."""

import numpy as np
import pytest

from oracle import hnsw as H


def _mk(metric, n=4000, d=48, conn=16, seed=5):
    rng = np.random.Generator(np.random.PCG64(seed))
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    ix = H.RefHnsw(d, metric=metric, connectivity=conn, expansion_add=64,
                   expansion_search=32, capacity=n, threads=4)
    ix.add(vecs)
    qs = rng.standard_normal((16, d), dtype=np.float32)
    return ix, qs


@pytest.mark.parametrize("metric", [H.METRIC_L2SQ, H.METRIC_IP, H.METRIC_COS])
def test_oracle_restatement_matches_usearch(metric):
    ix, qs = _mk(metric)
    ref_ids, ref_d = ix.search(qs, ef=32, k=10)
    g = ix.export_graph()
    assert g.adj[0][1].size > 0
    pos = 0
    for i in range(len(qs)):
        oi, od = H.oracle_search(g, qs[i], ef=32, k=10)
        pos += int((oi == ref_ids[i]).sum())
        np.testing.assert_allclose(od, ref_d[i], rtol=2e-5, atol=5e-5)
    # exact-position parity modulo rare FP tie flips
    assert pos >= len(qs) * 10 - 2


def test_export_graph_structure():
    ix, _ = _mk(H.METRIC_L2SQ, n=2000)
    g = ix.export_graph()
    n = len(g.levels)
    assert n == 2000
    assert g.levels[g.entry_slot] == g.max_level
    # every neighbor slot is valid; level-l adjacency only on nodes >= l
    for lvl in range(g.max_level + 1):
        offs, nbrs = g.adj[lvl]
        assert offs[-1] == nbrs.size
        if nbrs.size:
            assert nbrs.max() < n
        has = (offs[1:] - offs[:-1]) > 0
        assert not has[g.levels < lvl].any()


def test_ref_hnsw_search_recall():
    """usearch itself reaches high recall on clustered data (sanity that the
    shim drives it correctly)."""
    rng = np.random.Generator(np.random.PCG64(9))
    n, d = 8000, 32
    centers = rng.standard_normal((80, d), dtype=np.float32)
    comp = rng.integers(0, 80, n)
    vecs = centers[comp] + 0.4 * rng.standard_normal((n, d), dtype=np.float32)
    ix = H.RefHnsw(d, metric=H.METRIC_L2SQ, connectivity=16,
                   expansion_add=128, expansion_search=64, capacity=n,
                   threads=4)
    ix.add(vecs)
    qc = rng.integers(0, 80, 16)
    qs = centers[qc] + 0.4 * rng.standard_normal((16, d), dtype=np.float32)
    ids, _ = ix.search(qs, ef=64, k=10)
    from oracle import oracle as orc
    gt, _ = orc.bruteforce(orc.METRIC_L2SQ, vecs, qs, 10)
    recall = np.mean([len(set(ids[i]).intersection(gt[i])) / 10
                      for i in range(16)])
    assert recall > 0.9, recall


def test_filtered_oracle_matches_usearch():
    """Filtered search: the numpy restatement (predicate gates `top` only)
    matches usearch's own filtered_search on the same graph."""
    rng = np.random.Generator(np.random.PCG64(5))
    n, d = 4000, 48
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    ix = H.RefHnsw(d, metric=H.METRIC_L2SQ, connectivity=16, expansion_add=64,
                   expansion_search=32, capacity=n, threads=4)
    ix.add(vecs)
    qs = rng.standard_normal((12, d), dtype=np.float32)
    key_bits = np.zeros((n + 31) // 32, dtype=np.uint32)
    idxs = np.nonzero(rng.random(n) < 0.25)[0]
    np.bitwise_or.at(key_bits, idxs // 32,
                     (np.uint32(1) << (idxs % 32).astype(np.uint32)))
    ref_ids, _ = ix.search_filtered(qs, 32, 10, key_bits)
    g = ix.export_graph()
    aset = set(int(i) for i in idxs)
    pos = 0
    for i in range(12):
        oi, _ = H.oracle_search(g, qs[i], 32, 10, allowed=lambda kk: kk in aset)
        pos += int((oi == ref_ids[i]).sum())
    assert pos >= 118, pos
    valid = ref_ids[ref_ids >= 0]
    assert all(int(x) in aset for x in valid)


def test_native_usearch_file_loader():
    """The PRODUCT's usearch-file parser (matrixone_amd.hnsw.
    load_usearch_index — the bytes MO's hnsw/model.go persists) must
    reproduce exactly the graph the reference itself reports for the same
    saved file (shim export_graph): keys, levels, vectors, adjacency,
    entry point."""
    import tempfile

    from matrixone_amd.hnsw import load_usearch_index

    for metric, mname in [(H.METRIC_L2SQ, "l2sq"), (H.METRIC_COS, "cos"),
                          (H.METRIC_IP, "ip")]:
        ix, _ = _mk(metric, n=3000, d=40, conn=12, seed=metric + 2)
        g = ix.export_graph()
        with tempfile.NamedTemporaryFile(suffix=".usearch") as f:
            ix.save(f.name)
            mine = load_usearch_index(f.name)
        assert mine.metric == mname
        assert mine.dim == g.dim
        assert mine.entry_slot == g.entry_slot
        assert mine.max_level == g.max_level
        np.testing.assert_array_equal(mine.levels, g.levels)
        np.testing.assert_array_equal(mine.keys.astype(np.uint64),
                                      g.keys.astype(np.uint64))
        np.testing.assert_allclose(mine.vecs, g.vecs)
        assert len(mine.adj) == len(g.adj)
        for lvl, ((mo, mn), (ro, rn)) in enumerate(zip(mine.adj, g.adj)):
            np.testing.assert_array_equal(mo, ro, err_msg=f"offs l{lvl}")
            np.testing.assert_array_equal(mn, rn, err_msg=f"nbrs l{lvl}")


def test_native_loader_from_bytes():
    """Same parse from an in-memory byte blob (model chunks concatenated)."""
    import tempfile

    from matrixone_amd.hnsw import load_usearch_index

    ix, _ = _mk(H.METRIC_L2SQ, n=800, d=24, conn=8, seed=77)
    with tempfile.NamedTemporaryFile(suffix=".usearch") as f:
        ix.save(f.name)
        blob = open(f.name, "rb").read()
    g = ix.export_graph()
    mine = load_usearch_index(blob)
    np.testing.assert_array_equal(mine.levels, g.levels)
    np.testing.assert_allclose(mine.vecs, g.vecs)


def test_native_loader_rejects_garbage():
    import tempfile

    import pytest as _pytest

    from matrixone_amd.hnsw import load_usearch_index

    with _pytest.raises(ValueError):
        load_usearch_index(b"\x00" * 256)
    # truncated real file
    ix, _ = _mk(H.METRIC_L2SQ, n=500, d=16, conn=8, seed=3)
    with tempfile.NamedTemporaryFile(suffix=".usearch") as f:
        ix.save(f.name)
        blob = open(f.name, "rb").read()
    with _pytest.raises(Exception):
        load_usearch_index(blob[: len(blob) // 2])
