"""GPU tests for the round-2 ABI closure: the cuVS-shaped entries MatrixOne's
Go binding links that were named missing in round 1 —
extend/extend_float, delete_id (soft-delete bitset), add_chunk variants,
train_quantizer, save/load_file, info, search_quantize[_async], filter
columns + search_with_filter, and the usearchex-shaped HNSW entry.

Parity bar: results equal the oracle (or the equivalent already-parity-green
path) on the same inputs — bit-exact ids except FP ties, distances <= 1e-5
relative."""

import ctypes as ct
import os

import numpy as np
import pytest

import oracle.oracle as orc

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _mk(rng, n, d, nlist):
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    cents = vecs[rng.choice(n, nlist, replace=False)].copy()
    assign = ((vecs[:, None, :].astype(np.float64) -
               cents[None].astype(np.float64)) ** 2).sum(-1).argmin(1)
    return vecs, cents, assign


def _assert_close(ids, dists, rids, rdists, ctx=""):
    np.testing.assert_array_equal(ids, rids, err_msg=ctx)
    np.testing.assert_allclose(dists, rdists, rtol=2e-5, atol=2e-5,
                               err_msg=ctx)


def test_extend_matches_build_of_union():
    """gpu_ivf_flat_extend_float: rows join their nearest-centroid list
    (cuvs::ivf_flat::extend rule) — search == the oracle over the union
    corpus with the same assignment."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(900))
    n, extra, d, nlist, nq, probe, k = 6000, 700, 96, 32, 16, 6, 10
    vecs, cents, assign = _mk(rng, n + extra, d, nlist)
    base, ext = vecs[:n], vecs[n:]
    queries = rng.standard_normal((nq, d), dtype=np.float32)

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(base)
    ix.set_centroids(cents)
    ix.set_assignments(assign[:n].astype(np.int32))
    ix.build()
    ext_ids = np.arange(n, n + extra, dtype=np.int64)
    ix.extend(ext, ids=ext_ids)
    assert len(ix) == n + extra

    idx = orc.IvfIndex(cents, vecs, assign)
    ref_ids, ref_d = idx.search(orc.METRIC_L2SQ, queries, probe, k)
    ids, dists = ix.search(queries, k, probe)
    _assert_close(ids, dists, ref_ids, ref_d, "extend-union")

    # second extend (auto ids continue from the current length)
    more = rng.standard_normal((100, d), dtype=np.float32)
    ix.extend(more)
    assert len(ix) == n + extra + 100
    ix.close()


def test_extend_with_filter_and_submit():
    """filters and the pipelined submit path see extended rows too."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(901))
    n, extra, d, nlist, nq, probe, k = 4000, 500, 64, 16, 8, 5, 12
    vecs, cents, assign = _mk(rng, n + extra, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs[:n])
    ix.set_centroids(cents)
    ix.set_assignments(assign[:n].astype(np.int32))
    ix.build()
    ix.extend(vecs[n:], ids=np.arange(n, n + extra, dtype=np.int64))

    # filtered: allow only even ids (incl. extended ones)
    allowed = np.arange(0, n + extra, 2, dtype=np.int64)
    fids, fdists = ix.search_filtered(queries, k, probe,
                                      ix.filter_bitset(allowed))
    keep = np.zeros(n + extra, dtype=bool)
    keep[allowed] = True
    sub = np.where(keep)[0]
    idx = orc.IvfIndex(cents, vecs[sub], assign[sub], ids=sub.astype(np.int64))
    rids, rdists = idx.search(orc.METRIC_L2SQ, queries, probe, k)
    _assert_close(fids, fdists, rids, rdists, "extend+filter")

    # pipelined submit sees the tail as well
    import torch
    qt = torch.from_numpy(queries).cuda()
    t1 = ix.search_submit(qt, k, probe)
    sids, sdists = ix.search_collect(t1)
    eids, edists = ix.search(queries, k, probe)
    _assert_close(sids, sdists, eids, edists, "extend+submit")
    ix.close()


def test_delete_id_soft_delete():
    """gpu_ivf_flat_delete_id: deleted ids vanish from every search variant;
    unknown ids are a silent no-op (index_base.hpp:1142-1167)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(902))
    n, d, nlist, nq, probe, k = 5000, 64, 16, 12, 6, 10
    vecs, cents, assign = _mk(rng, n, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()

    base_ids, _ = ix.search(queries, k, probe)
    victims = sorted(set(int(x) for x in base_ids[:, 0] if x >= 0))[:8]
    for v in victims:
        ix.delete_id(v)
    ix.delete_id(10 ** 12)  # unknown: silent no-op
    assert ix.info()["deleted_count"] == len(victims)

    ids, dists = ix.search(queries, k, probe)
    assert not (set(victims) & set(ids.ravel().tolist()))
    keep = np.ones(n, dtype=bool)
    keep[victims] = False
    sub = np.where(keep)[0]
    idx = orc.IvfIndex(cents, vecs[sub], assign[sub], ids=sub.astype(np.int64))
    rids, rdists = idx.search(orc.METRIC_L2SQ, queries, probe, k)
    _assert_close(ids, dists, rids, rdists, "delete")

    # deletes compose with a user filter
    allowed = np.arange(0, n, 3, dtype=np.int64)
    fids, fdists = ix.search_filtered(queries, k, probe,
                                      ix.filter_bitset(allowed))
    keep2 = np.zeros(n, dtype=bool)
    keep2[allowed] = True
    keep2[victims] = False
    sub2 = np.where(keep2)[0]
    idx2 = orc.IvfIndex(cents, vecs[sub2], assign[sub2],
                        ids=sub2.astype(np.int64))
    r2, rd2 = idx2.search(orc.METRIC_L2SQ, queries, probe, k)
    _assert_close(fids, fdists, r2, rd2, "delete+filter")

    # delete an EXTENDED row
    ext = rng.standard_normal((50, d), dtype=np.float32)
    ix.extend(ext, ids=np.arange(n, n + 50, dtype=np.int64))
    ix.delete_id(n + 3)
    ids3, _ = ix.search(queries, max(k, 32), probe)
    assert (n + 3) not in set(ids3.ravel().tolist())
    ix.close()


def test_save_load_file_round_trip(tmp_path):
    """gpu_ivf_flat_save + gpu_ivf_flat_load_file restore an index with
    extends AND soft deletes folded in."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(903))
    n, d, nlist, nq, probe, k = 4000, 48, 16, 10, 5, 10
    vecs, cents, assign = _mk(rng, n, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n - 300)
    ix.add(vecs[:n - 300])
    ix.set_centroids(cents)
    ix.set_assignments(assign[:n - 300].astype(np.int32))
    ix.build()
    ix.extend(vecs[n - 300:],
              ids=np.arange(n - 300, n, dtype=np.int64))
    for v in (5, 17, n - 100):
        ix.delete_id(v)
    want_ids, want_d = ix.search(queries, k, probe)
    path = str(tmp_path / "index.moann")
    ix.save_file(path)
    ix.close()

    L = engine.lib()
    err = engine._Err()
    bp = engine.BuildParams(nlist, True, 0.5)
    dev = (ct.c_int * 1)(0)
    h = L.gpu_ivf_flat_load_file(path.encode(), d, engine.METRICS["l2sq"],
                                 bp, dev, 1, 0, 0, 0, 0, err.ref)
    err.check("load_file")
    assert h
    ix2 = engine.IvfFlatIndex.__new__(engine.IvfFlatIndex)
    ix2.dim, ix2.nlist, ix2.metric = d, nlist, "l2sq"
    ix2.device, ix2.btype = 0, "f32"
    ix2._h = h
    assert len(ix2) == n
    assert ix2.info()["deleted_count"] == 3
    got_ids, got_d = ix2.search(queries, k, probe)
    _assert_close(got_ids, got_d, want_ids, want_d, "save/load_file")
    ix2.close()


def test_storage_typed_search_and_add_int8():
    """gpu_ivf_flat_search / add_chunk take STORAGE-typed (int8) payloads:
    byte queries == search_quantize of the dequantized f32 queries; byte
    adds == float adds of the dequantized rows."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(904))
    n, d, nlist, nq, probe, k = 4000, 64, 16, 8, 6, 10
    vecs, cents, assign = _mk(rng, n, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    lo, hi = float(vecs.min()), float(vecs.max())

    def q_int8(x, mul, add):
        v = np.round((x * mul).astype(np.float32) + np.float32(add))
        return np.clip(v, -128, 127).astype(np.int8)

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                             qtype="int8")
    ix.set_quantizer(lo, hi)
    mn = ct.c_float()
    mx = ct.c_float()
    err = engine._Err()
    engine.lib().gpu_ivf_flat_get_quantizer(ix._h, ct.byref(mn),
                                            ct.byref(mx), err.ref)
    err.check("get_quantizer")
    rng_ = hi - lo
    mul = 255.0 / rng_
    add = -lo * mul - 128.0
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    want_ids, want_d = ix.search(queries, k, probe)  # base-typed f32 path

    qbytes = q_int8(queries, mul, add)
    res = engine.lib().gpu_ivf_flat_search(
        ix._h, qbytes.ctypes.data, nq, d, k, engine.SearchParams(probe),
        engine._Err().ref)
    got_ids, got_d = ix._read_result(res, nq, k, "storage search")
    # byte queries carry quantization error vs the f32 queries; compare
    # against search_quantize of the DEQUANTIZED byte queries instead
    deq = (qbytes.astype(np.float64) - add) / mul
    eq_ids, eq_d = ix.search(deq.astype(np.float32), k, probe)
    _assert_close(got_ids, got_d, eq_ids, eq_d, "storage-typed search")

    # storage-typed add_chunk reproduces the float-add index exactly
    ix3 = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                              qtype="int8")
    ix3.set_quantizer(lo, hi)
    vbytes = q_int8(vecs, mul, add)
    err3 = engine._Err()
    engine.lib().gpu_ivf_flat_add_chunk(ix3._h, vbytes.ctypes.data, n, None,
                                        err3.ref)
    err3.check("add_chunk bytes")
    ix3.set_centroids(cents)
    ix3.set_assignments(assign.astype(np.int32))
    ix3.build()
    b_ids, b_d = ix3.search(queries, k, probe)
    _assert_close(b_ids, b_d, want_ids, want_d, "storage-typed add")
    ix3.close()
    ix.close()


def test_f16_base_combo():
    """btype F16 (ivf_flat_c.h:33-35): half queries/rows through
    add_chunk_quantize + search_quantize equal an f32-base index over the
    decoded values (the half->f32 cast is exact)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(905))
    n, d, nlist, nq, probe, k = 3000, 64, 16, 8, 5, 10
    vecs, cents, assign = _mk(rng, n, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    vecs_h = vecs.astype(np.float16)
    queries_h = queries.astype(np.float16)

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                             qtype="f16", btype="f16")
    err = engine._Err()
    engine.lib().gpu_ivf_flat_add_chunk_quantize(
        ix._h, vecs_h.ctypes.data, n, None, err.ref)
    err.check("add_chunk_quantize half")
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    err2 = engine._Err()
    res = engine.lib().gpu_ivf_flat_search_quantize(
        ix._h, queries_h.ctypes.data, nq, d, k, engine.SearchParams(probe),
        err2.ref)
    err2.check("search_quantize half")
    got_ids, got_d = ix._read_result(res, nq, k, "f16 base")
    ix.close()

    ix2 = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                              qtype="f16")
    ix2.add(vecs_h.astype(np.float32))
    ix2.set_centroids(cents)
    ix2.set_assignments(assign.astype(np.int32))
    ix2.build()
    want_ids, want_d = ix2.search(queries_h.astype(np.float32), k, probe)
    _assert_close(got_ids, got_d, want_ids, want_d, "f16 base == f32 base")
    ix2.close()


def test_filter_columns_and_search_with_filter():
    """set_filter_columns + add_filter_chunk + search_with_filter ==
    the (already parity-pinned) per-call moann pred-filtered path on the
    same predicates (cagra_c.h:163-199 surface)."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(906))
    n, d, nlist, nq, probe, k = 5000, 48, 16, 8, 6, 12
    vecs, cents, assign = _mk(rng, n, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    price = rng.integers(0, 1000, n).astype(np.int64)
    nulls = np.zeros((n + 31) // 32, dtype=np.uint32)
    for i in range(0, n, 97):  # some NULLs
        nulls[i // 32] |= np.uint32(1) << (i % 32)

    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n)
    ix.set_filter_columns('[{"name":"price","type":1}]', n)
    half = n // 2
    ix.add_filter_chunk(0, price[:half],
                        nulls[: (half + 31) // 32])  # chunked adds
    tailbits = np.zeros((n - half + 31) // 32, dtype=np.uint32)
    for i in range(half, n):
        if (nulls[i // 32] >> (i % 32)) & 1:
            j = i - half
            tailbits[j // 32] |= np.uint32(1) << (j % 32)
    ix.add_filter_chunk(0, price[half:], tailbits)
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()

    preds = '[{"col":0,"op":"between","lo":100,"hi":600}]'
    gids, gdists = ix.search_with_filter(queries, k, probe, preds)
    eids, edists = ix.search_pred_filtered(
        queries, k, probe, [("price", engine.COL_I64)], [price], preds,
        nulls=[nulls])
    np.testing.assert_array_equal(gids, eids)
    np.testing.assert_allclose(gdists, edists, rtol=2e-5, atol=2e-5)

    # empty predicate == unfiltered
    uids, udists = ix.search_with_filter(queries, k, probe, "")
    pids, pdists = ix.search(queries, k, probe)
    np.testing.assert_array_equal(uids, pids)

    # async filtered variant returns via search_wait
    err = engine._Err()
    job = engine.lib().gpu_ivf_flat_search_quantize_with_filter_async(
        ix._h, queries.ctypes.data, nq, d, k, engine.SearchParams(probe),
        preds.encode(), err.ref)
    err.check("filtered async")
    assert job
    ix._async_shape = (nq, k)
    aids, adists = ix.search_wait(job)
    np.testing.assert_array_equal(aids, gids)
    ix.close()


def test_train_quantizer():
    """train_quantizer computes the central-99% quantile range; the stored
    quantizer reproduces it through get_quantizer."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(907))
    n, d, nlist = 3000, 32, 8
    vecs, cents, assign = _mk(rng, n, d, nlist)
    ix = engine.IvfFlatIndex(d, nlist, metric="l2sq", capacity=n,
                             qtype="int8")
    ix.train_quantizer(vecs)
    mn = ct.c_float()
    mx = ct.c_float()
    err = engine._Err()
    engine.lib().gpu_ivf_flat_get_quantizer(ix._h, ct.byref(mn),
                                            ct.byref(mx), err.ref)
    err.check("get_quantizer")
    lo = np.quantile(vecs.ravel(), 0.005)
    hi = np.quantile(vecs.ravel(), 0.995)
    assert abs(mn.value - lo) < 0.05 and abs(mx.value - hi) < 0.05
    # trains an affine map good enough to search with
    ix.add(vecs)
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    q = rng.standard_normal((4, d), dtype=np.float32)
    ids, _ = ix.search(q, 5, 4)
    assert (ids >= 0).all()
    ix.close()


def test_usearchex_membership_entry_gpu():
    """usearchex_filtered_search_with_membership == the existing (pinned)
    moann_hnsw_search_filtered with the docfilter-evaluated bitset, on a
    small synthetic graph."""
    from matrixone_amd import engine
    from matrixone_amd import hnsw as mh
    rng = np.random.Generator(np.random.PCG64(908))
    n, d, k = 2000, 32, 10
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    keys = (np.arange(n, dtype=np.int64) * 7 + 1)

    # tiny usearch-shaped graph: exact kNN adjacency (M=8), level-0 only
    M = 8
    d2 = ((vecs[:, None, :16] - vecs[None, :, :16]) ** 2).sum(-1)
    np.fill_diagonal(d2, np.inf)
    nbrs = np.argsort(d2, axis=1)[:, :M].astype(np.uint32)
    offs = (np.arange(n + 1, dtype=np.int64) * M)

    L = mh._bind(engine.lib())
    err = engine._Err()
    h = L.moann_hnsw_new(d, 0, n, 0, 0, (ct.c_int * 1)(0), 1, err.ref)
    err.check("hnsw new")
    for call, args in (
        (L.moann_hnsw_set_vectors, (h, vecs.ctypes.data)),
        (L.moann_hnsw_set_keys, (h, keys.ctypes.data)),
    ):
        e = engine._Err()
        call(*args, e.ref)
        e.check("hnsw setup")
    e = engine._Err()
    L.moann_hnsw_set_level0(h, offs.ctypes.data, nbrs.ctypes.data,
                            nbrs.size, e.ref)
    e.check("level0")
    e = engine._Err()
    L.moann_hnsw_build(h, e.ref)
    e.check("build")
    e = engine._Err()
    L.moann_hnsw_set_ef(h, 64, e.ref)
    e.check("set_ef")

    # membership filter: sorted64 over a subset of keys
    allowed = np.sort(keys[rng.choice(n, n // 3, replace=False)]).astype(
        np.uint64)
    payload = np.concatenate([np.array([len(allowed)], dtype=np.uint64),
                              allowed]).tobytes()
    e = engine._Err()
    fh = L.moann_docfilter_parse(4, payload, len(payload), e.ref)
    e.check("parse")

    q = rng.standard_normal(d, dtype=np.float32)
    out_keys = np.full(k, -1, dtype=np.int64)
    out_d = np.full(k, np.float32(np.inf), dtype=np.float32)
    errp = ct.c_char_p()
    found = L.usearchex_filtered_search_with_membership(
        h, q.ctypes.data, 1, k, fh, 4, out_keys.ctypes.data,
        out_d.ctypes.data, ct.byref(errp))
    assert errp.value is None, errp.value
    assert found > 0
    allowed_set = set(int(x) for x in allowed)
    assert all(int(x) in allowed_set for x in out_keys[:found])

    # equivalence vs the explicit-bitset filtered entry
    words = np.zeros((n + 31) // 32, dtype=np.uint32)
    member = np.isin(keys, allowed.astype(np.int64))
    idxs = np.where(member)[0]
    np.bitwise_or.at(words, idxs // 32,
                     (np.uint32(1) << (idxs % 32).astype(np.uint32)))
    ref_keys = np.full(k, -1, dtype=np.int64)
    ref_d = np.full(k, np.float32(np.inf), dtype=np.float32)
    e = engine._Err()
    L.moann_hnsw_search_filtered(h, q.ctypes.data, 1, d, 64, k,
                                 words.ctypes.data, ref_keys.ctypes.data,
                                 ref_d.ctypes.data, e.ref)
    e.check("filtered ref")
    np.testing.assert_array_equal(out_keys[:found], ref_keys[:found])
    np.testing.assert_allclose(out_d[:found], ref_d[:found], rtol=1e-6)

    # unknown scalar kind fails loudly through the error out-param
    errp2 = ct.c_char_p()
    got = L.usearchex_filtered_search_with_membership(
        h, q.ctypes.data, 2, k, fh, 4, out_keys.ctypes.data,
        out_d.ctypes.data, ct.byref(errp2))
    assert got == 0 and errp2.value

    L.moann_docfilter_free(fh)
    e = engine._Err()
    L.moann_hnsw_destroy(h, e.ref)
    e.check("destroy")


def test_dataset_ctor():
    """gpu_ivf_flat_new == new_empty + add_chunk of the whole dataset."""
    from matrixone_amd import engine
    rng = np.random.Generator(np.random.PCG64(909))
    n, d, nlist, nq, probe, k = 3000, 32, 8, 6, 4, 8
    vecs, cents, assign = _mk(rng, n, d, nlist)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    L = engine.lib()
    err = engine._Err()
    bp = engine.BuildParams(nlist, True, 0.5)
    dev = (ct.c_int * 1)(0)
    h = L.gpu_ivf_flat_new(vecs.ctypes.data, n, d, engine.METRICS["l2sq"],
                           bp, dev, 1, 0, 0, 0, 0, None, err.ref)
    err.check("new")
    assert h
    ix = engine.IvfFlatIndex.__new__(engine.IvfFlatIndex)
    ix.dim, ix.nlist, ix.metric, ix.device, ix.btype = d, nlist, "l2sq", 0, "f32"
    ix._h = h
    ix.set_centroids(cents)
    ix.set_assignments(assign.astype(np.int32))
    ix.build()
    got_ids, got_d = ix.search(queries, k, probe)
    idx = orc.IvfIndex(cents, vecs, assign)
    rids, rd = idx.search(orc.METRIC_L2SQ, queries, probe, k)
    _assert_close(got_ids, got_d, rids, rd, "dataset ctor")
    ix.close()
