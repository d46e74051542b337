"""GPU parity: the CDNA4 HNSW beam-search kernel vs the numpy oracle
restatement AND usearch's own results, all on the SAME exported graph.
Requires a real MI355X (`-m gpu`); builds the usearch graph live via the
prebuilt oracle/_ref/libref_hnsw.so that travels with the repo."""

import numpy as np
import pytest

from oracle import hnsw as H

pytestmark = pytest.mark.gpu


def _mk(metric, n=6000, d=96, conn=16, seed=11, nq=24):
    rng = np.random.Generator(np.random.PCG64(seed))
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    ix = H.RefHnsw(d, metric=metric, connectivity=conn, expansion_add=64,
                   expansion_search=48, capacity=n, threads=4)
    ix.add(vecs)
    qs = rng.standard_normal((nq, d), dtype=np.float32)
    return ix, qs


def _graph_data(g):
    from matrixone_amd.hnsw import HnswGraphData
    mname = {H.METRIC_L2SQ: "l2sq", H.METRIC_IP: "ip", H.METRIC_COS: "cos"}
    return HnswGraphData(dim=g.dim, metric=mname[g.metric],
                         entry_slot=g.entry_slot, max_level=g.max_level,
                         levels=g.levels, keys=g.keys, vecs=g.vecs, adj=g.adj)


@pytest.mark.parametrize("metric", [H.METRIC_L2SQ, H.METRIC_IP, H.METRIC_COS])
def test_hnsw_gpu_vs_oracle(metric):
    from matrixone_amd.hnsw import HnswIndex
    ix, qs = _mk(metric)
    g = ix.export_graph()
    gpu = HnswIndex(_graph_data(g))
    k, ef = 10, 48
    gids, gdists = gpu.search(qs, ef, k)
    ref_ids, ref_d = ix.search(qs, ef=ef, k=k)
    mismatch_pos = 0
    for i in range(len(qs)):
        oi, od = H.oracle_search(g, qs[i], ef=ef, k=k)
        # GPU vs oracle: same algorithm, same graph — ids equal except FP ties
        for j in range(k):
            if gids[i, j] != oi[j]:
                assert abs(float(gdists[i, j]) - float(od[j])) <= 1e-5 * max(
                    1e-30, abs(float(od[j]))) + 1e-5, (
                    f"metric={metric} q={i} j={j}: {gids[i, j]} vs {oi[j]} "
                    f"({gdists[i, j]} vs {od[j]})")
        np.testing.assert_allclose(gdists[i], od, rtol=2e-5, atol=5e-5)
        mismatch_pos += int((gids[i] != ref_ids[i]).sum())
    # GPU vs usearch direct: near-total position agreement
    assert mismatch_pos <= len(qs) * k * 0.05, mismatch_pos
    gpu.close()


def test_hnsw_gpu_various_ef_k():
    from matrixone_amd.hnsw import HnswIndex
    ix, qs = _mk(H.METRIC_L2SQ, n=3000, d=33)  # dim not multiple of 4
    g = ix.export_graph()
    gpu = HnswIndex(_graph_data(g))
    for ef, k in [(8, 8), (64, 1), (128, 100), (256, 256)]:
        gids, gdists = gpu.search(qs[:6], ef, k)
        for i in range(6):
            oi, od = H.oracle_search(g, qs[i], ef=max(ef, k), k=k)
            np.testing.assert_allclose(gdists[i], od, rtol=2e-5, atol=5e-5,
                                       err_msg=f"ef={ef} k={k} q={i}")
    gpu.close()


def test_hnsw_host_mirror_transform():
    """HnswSearch applies DistanceTransformHnsw (ip: -1; l2: sqrt)."""
    from matrixone_amd.hnsw import HnswSearch
    from matrixone_amd.ivfflat import RuntimeConfig
    ix, qs = _mk(H.METRIC_IP, n=2000, d=32)
    g = ix.export_graph()
    hs = HnswSearch(_graph_data(g), op_type="vector_ip_ops", ef_search=48)
    hs.Load()
    ids, d64 = hs.Search(None, qs[0],
                         RuntimeConfig(limit=5, orig_func_name="inner_product"))
    raw_ids, raw_d = hs.index.search(qs[:1], 48, 5)
    np.testing.assert_array_equal(ids, raw_ids)
    np.testing.assert_allclose(d64[0], raw_d[0].astype(np.float64) - 1.0)
    hs.Destroy()


def test_hnsw_keys_mapping():
    """Custom usearch keys (MO pks) come back, not slots."""
    from matrixone_amd.hnsw import HnswIndex
    rng = np.random.Generator(np.random.PCG64(3))
    n, d = 1500, 16
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    keys = (np.arange(n, dtype=np.uint64) * 7 + 1000)
    ix = H.RefHnsw(d, metric=H.METRIC_L2SQ, connectivity=8, expansion_add=32,
                   expansion_search=32, capacity=n, threads=1)
    ix.add(vecs, keys=keys)
    g = ix.export_graph()
    gpu = HnswIndex(_graph_data(g))
    qs = rng.standard_normal((4, d), dtype=np.float32)
    gids, _ = gpu.search(qs, 32, 5)
    ref_ids, _ = ix.search(qs, ef=32, k=5)
    valid = gids >= 0
    assert ((gids[valid] - 1000) % 7 == 0).all()
    assert (gids == ref_ids).mean() > 0.9
    gpu.close()


def test_hnsw_filtered_gpu():
    """Filtered GPU beam (§8a4): strict parity vs its spec (the batch-radius
    model in oracle/hnsw.py), plus filter validity, ~0.9 id agreement and
    recall EQUIVALENCE vs usearch's own filtered_search."""
    from matrixone_amd.hnsw import HnswIndex
    from oracle import oracle as orc
    ix, qs = _mk(H.METRIC_L2SQ, n=5000, d=64)
    g = ix.export_graph()
    rng = np.random.Generator(np.random.PCG64(77))
    n = len(g.levels)
    key_bits = np.zeros((n + 31) // 32, dtype=np.uint32)
    idxs = np.nonzero(rng.random(n) < 0.3)[0]
    np.bitwise_or.at(key_bits, idxs // 32,
                     (np.uint32(1) << (idxs % 32).astype(np.uint32)))
    slot_bits = np.zeros((n + 31) // 32, dtype=np.uint32)
    pass_slots = np.nonzero(np.isin(g.keys.astype(np.int64), idxs))[0]
    np.bitwise_or.at(slot_bits, pass_slots // 32,
                     (np.uint32(1) << (pass_slots % 32).astype(np.uint32)))
    gpu = HnswIndex(_graph_data(g))
    ef, k = 48, 10
    gids, gdists = gpu.search_filtered(qs, ef, k, slot_bits)
    aset = set(int(i) for i in idxs)
    assert all(int(x) in aset for x in gids[gids >= 0]), "filtered-out id"

    # 1. strict vs the spec model (same algorithm: ids equal modulo FP ties)
    spec_pos = 0
    for i in range(len(qs)):
        mi, md = H.oracle_search_filtered_batch(
            g, qs[i], ef, k, lambda kk: kk in aset)
        for j in range(k):
            if gids[i, j] != mi[j]:
                assert abs(float(gdists[i, j]) - float(md[j])) <=                     2e-5 * max(1e-30, abs(float(md[j]))) + 2e-5, (i, j)
        spec_pos += int((gids[i] == mi).sum())
    assert spec_pos >= len(qs) * k - 4, spec_pos

    # 2. vs usearch: documented approximation band + recall equivalence
    ref_ids, _ = ix.search_filtered(qs, ef, k, key_bits)
    assert (gids == ref_ids).mean() > 0.85
    allowed_rows = np.array(sorted(aset))
    vecs_allowed = g.vecs[np.isin(g.keys.astype(np.int64), idxs)]
    keys_allowed = g.keys[np.isin(g.keys.astype(np.int64), idxs)].astype(np.int64)
    gt, _ = orc.bruteforce(orc.METRIC_L2SQ, vecs_allowed, qs, k,
                           ids=keys_allowed)
    def recall(res):
        return np.mean([len(set(res[i]).intersection(gt[i])) / k
                        for i in range(len(qs))])
    r_gpu, r_ref = recall(gids), recall(ref_ids)
    assert r_gpu >= r_ref - 0.03, (r_gpu, r_ref)

    u_ids, _ = gpu.search(qs, ef, k)
    ru_ids, _ = ix.search(qs, ef=ef, k=k)
    assert (u_ids == ru_ids).mean() > 0.95
    gpu.close()


def test_multi_model_fanout():
    """Multi-model search (hnsw/search.go:89-183): merged results over two
    disjoint mini-models equal usearch over each model merged the same way."""
    from matrixone_amd.hnsw import MultiModelHnswSearch
    from matrixone_amd.ivfflat import RuntimeConfig
    rng = np.random.Generator(np.random.PCG64(21))
    d, nq, k = 48, 8, 10
    graphs, refs = [], []
    for m in range(2):
        vecs = rng.standard_normal((2500, d), dtype=np.float32)
        keys = np.arange(2500, dtype=np.uint64) + m * 1_000_000
        ix = H.RefHnsw(d, metric=H.METRIC_L2SQ, connectivity=16,
                       expansion_add=64, expansion_search=48, capacity=2500,
                       threads=2)
        ix.add(vecs, keys=keys)
        graphs.append(_graph_data(ix.export_graph()))
        refs.append(ix)
    qs = rng.standard_normal((nq, d), dtype=np.float32)
    mm = MultiModelHnswSearch(graphs, op_type="vector_l2sq_ops", ef_search=48)
    mm.Load()
    ids, d64 = mm.Search(None, qs, RuntimeConfig(limit=k,
                                                 orig_func_name="l2_distance_sq"))
    # reference merge of per-model usearch results
    bi, bd = [], []
    for ix in refs:
        ri, rd = ix.search(qs, ef=48, k=k)
        bi.append(ri)
        bd.append(rd)
    from oracle import oracle as orc
    mi, md = orc.topk_merge(np.stack(bi), np.stack(bd).astype(np.float32), k)
    agree = (ids == mi).mean()
    assert agree > 0.95, agree
    np.testing.assert_allclose(d64, md.astype(np.float64), rtol=3e-5, atol=3e-5)
    mm.Destroy()
