"""Oracle self-checks: known-answer tests from the reference's own test
vectors, golden-fixture pinning, and agreement with the reference's compiled
arithmetic (usearch/SimSIMD exact search).

All CPU-only (`-m "not gpu"`).
"""

import glob
import os

import numpy as np
import pytest

from oracle import oracle as orc

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


# KATs lifted from pkg/vectorindex/metric/distance_func_test.go:155-232 etc.
L2_KATS = [
    ([1, 2, 3, 4], [1, 2, 4, 5], 1.4142135623730951),
    ([10, 20, 30, 40], [10.5, 21.5, 31.5, 43.5], 4.123105625617661),
    ([1, 1], [4, 1], 3.0),
    ([4, 1], [1, 4], 4.242640687119285),
    ([1, 2, 3, 4, 5, 6, 7, 8, 9, 10], [2, 3, 4, 5, 6, 7, 8, 9, 10, 11],
     3.1622776601683795),
    ([1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 1, 2, 3, 4,
      5, 6, 7],
     [2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 2, 3, 4,
      5, 6, 7, 8], 5.196152422706632),
]

COS_KATS = [
    ([1, 2, 3, 4], [1, 2, 4, 5], 0.003993481192393733),
    ([10, 20, 30, 40], [10.5, 21.5, 31.5, 43.5], 0.0001253573895874105),
    ([1, 1], [4, 1], 0.1425070742874559),
    ([4, 1], [1, 4], 0.5294117647058824),
]


@pytest.mark.parametrize("v1,v2,want", L2_KATS)
def test_l2_kat(v1, v2, want):
    a = np.array(v1, dtype=np.float32)
    b = np.array(v2, dtype=np.float32)
    got = orc.distance(orc.METRIC_L2, a, b)
    assert got == pytest.approx(want, rel=1e-6)
    got_sq = orc.distance(orc.METRIC_L2SQ, a, b)
    assert got_sq == pytest.approx(want * want, rel=1e-6)


@pytest.mark.parametrize("v1,v2,want", COS_KATS)
def test_cos_kat(v1, v2, want):
    a = np.array(v1, dtype=np.float32)
    b = np.array(v2, dtype=np.float32)
    assert orc.distance(orc.METRIC_COS, a, b) == pytest.approx(want, rel=1e-5, abs=1e-7)


def test_ip_convention():
    # distance_func.go:207 — InnerProduct returns MINUS the dot product
    a = np.array([1, 2, 3], dtype=np.float32)
    b = np.array([4, 5, 6], dtype=np.float32)
    assert orc.distance(orc.METRIC_IP, a, b) == pytest.approx(-32.0)


def test_cos_zero_vector():
    # distance_func.go:270 — zero-magnitude vector -> distance 1.0
    a = np.zeros(4, dtype=np.float32)
    b = np.array([1, 2, 3, 4], dtype=np.float32)
    assert orc.distance(orc.METRIC_COS, a, b) == 1.0


def test_distance_vs_numpy_f64():
    rng = np.random.Generator(np.random.PCG64(7))
    for d in (1, 7, 8, 9, 31, 128, 768, 1536):
        a = rng.standard_normal(d).astype(np.float32)
        b = rng.standard_normal(d).astype(np.float32)
        ref = ((a.astype(np.float64) - b.astype(np.float64)) ** 2).sum()
        assert orc.distance(orc.METRIC_L2SQ, a, b) == pytest.approx(ref, rel=1e-5)
        ref_ip = -(a.astype(np.float64) * b.astype(np.float64)).sum()
        assert orc.distance(orc.METRIC_IP, a, b) == pytest.approx(ref_ip, rel=1e-4, abs=1e-5)


def _ids_match_with_tie_tolerance(ids_a, dists_a, ids_b, dists_b, rtol=1e-5):
    """Neighbor ids must match position by position, except where the
    distances at the mismatching positions are equal within rtol (FP tie
    order is unspecified in the reference heap)."""
    for q in range(ids_a.shape[0]):
        for j in range(ids_a.shape[1]):
            if ids_a[q, j] == ids_b[q, j]:
                continue
            da, db = dists_a[q, j], dists_b[q, j]
            denom = max(abs(da), abs(db), 1e-30)
            assert abs(da - db) / denom < rtol, (
                f"q={q} j={j}: id {ids_a[q, j]} vs {ids_b[q, j]} "
                f"dist {da} vs {db}")


def test_golden_ivf_fixtures():
    cases = sorted(glob.glob(os.path.join(GOLDEN, "ivf_*.npz")))
    assert cases, "golden fixtures missing — run tests/golden/make_golden.py"
    for path in cases:
        z = np.load(path)
        idx = orc.IvfIndex(z["centroids"], z["vecs"], z["assign"], ids=z["ids"])
        out_ids, out_dists, out_probe = idx.search(
            int(z["metric"]), z["queries"], int(z["probe"]), int(z["k"]),
            orig_l2=bool(z["orig_l2"]), want_probe=True)
        np.testing.assert_array_equal(out_ids, z["out_ids"], err_msg=path)
        np.testing.assert_allclose(out_dists, z["out_dists"], rtol=0, atol=0,
                                   err_msg=path)
        np.testing.assert_array_equal(out_probe, z["out_probe"], err_msg=path)


def test_golden_bf_fixtures():
    cases = sorted(glob.glob(os.path.join(GOLDEN, "bf_*.npz")))
    assert cases
    for path in cases:
        z = np.load(path)
        out_ids, out_dists = orc.bruteforce(
            int(z["metric"]), z["base"], z["queries"], int(z["k"]),
            orig_l2=bool(z["orig_l2"]))
        np.testing.assert_array_equal(out_ids, z["out_ids"], err_msg=path)
        np.testing.assert_array_equal(out_dists, z["out_dists"], err_msg=path)


def test_bf_vs_usearch_fixture():
    """Oracle brute force vs the reference's own compiled arithmetic
    (usearch_exact_search results stored in the fixtures). usearch's IP
    metric is 1 - a.b vs MO's -a.b (metric/types.go:219-243 transform), so
    shift by 1 before comparing."""
    found = 0
    for path in sorted(glob.glob(os.path.join(GOLDEN, "bf_*.npz"))):
        z = np.load(path)
        if "usearch_keys" not in z:
            continue
        found += 1
        metric = int(z["metric"])
        ud = z["usearch_dists"].astype(np.float64)
        if metric == orc.METRIC_IP:
            ud = ud - 1.0  # DistanceTransformHnsw (metric/types.go:241)
        _ids_match_with_tie_tolerance(
            z["out_ids"], z["out_dists"].astype(np.float64),
            z["usearch_keys"].astype(np.int64), ud, rtol=2e-5)
        np.testing.assert_allclose(z["out_dists"].astype(np.float64), ud,
                                   rtol=2e-5, atol=2e-5, err_msg=path)
    assert found >= 3, "fixtures were generated without usearch pinning"


def test_bf_vs_usearch_live():
    """Re-derive the agreement live when _ref is buildable (this container and
    the GPU box, where the prebuilt .so travels)."""
    if orc.usearch_lib() is None:
        pytest.skip("_ref/libusearch_c.so not available")
    rng = np.random.Generator(np.random.PCG64(11))
    base = rng.standard_normal((3000, 96), dtype=np.float32)
    queries = rng.standard_normal((8, 96), dtype=np.float32)
    ok, od = orc.bruteforce(orc.METRIC_L2SQ, base, queries, 12)
    uk, ud = orc.usearch_exact_search(base, queries, 12, metric="l2sq")
    _ids_match_with_tie_tolerance(ok, od.astype(np.float64),
                                  uk.astype(np.int64), ud.astype(np.float64))


def test_ivf_probe_covers_bruteforce_at_full_probe():
    """probe == nlist makes IVF exact: must equal brute force."""
    rng = np.random.Generator(np.random.PCG64(13))
    vecs = rng.standard_normal((1200, 24), dtype=np.float32)
    cents = rng.standard_normal((8, 24), dtype=np.float32)
    assign = ((vecs[:, None, :] - cents[None]) ** 2).sum(-1).argmin(1)
    idx = orc.IvfIndex(cents, vecs, assign)
    qs = rng.standard_normal((6, 24), dtype=np.float32)
    ivf_ids, ivf_d = idx.search(orc.METRIC_L2SQ, qs, probe=8, k=15)
    bf_ids, bf_d = orc.bruteforce(orc.METRIC_L2SQ, vecs, qs, 15)
    np.testing.assert_array_equal(ivf_ids, bf_ids)
    np.testing.assert_array_equal(ivf_d, bf_d)


def test_empty_and_padding():
    """k > candidates: pad with (-1, FLT_MAX) — cgo/cuvs/helper.h:137-167."""
    vecs = np.ones((3, 4), dtype=np.float32)
    cents = np.zeros((2, 4), dtype=np.float32)
    cents[1] += 100
    assign = np.zeros(3, dtype=np.int64)
    idx = orc.IvfIndex(cents, vecs, assign)
    q = np.zeros((1, 4), dtype=np.float32)
    ids, dists = idx.search(orc.METRIC_L2SQ, q, probe=1, k=8)
    assert (ids[0, :3] >= 0).all()
    assert (ids[0, 3:] == -1).all()
    assert np.all(dists[0, 3:] == np.finfo(np.float32).max)


def test_l2_sqrt_transform():
    """orig l2_distance over an l2sq index applies sqrt at the end
    (metric/types.go:245-251)."""
    rng = np.random.Generator(np.random.PCG64(17))
    vecs = rng.standard_normal((500, 16), dtype=np.float32)
    cents = vecs[:4].copy()
    assign = ((vecs[:, None, :] - cents[None]) ** 2).sum(-1).argmin(1)
    idx = orc.IvfIndex(cents, vecs, assign)
    q = rng.standard_normal((3, 16), dtype=np.float32)
    ids_sq, d_sq = idx.search(orc.METRIC_L2SQ, q, probe=4, k=5)
    ids_l2, d_l2 = idx.search(orc.METRIC_L2, q, probe=4, k=5, orig_l2=True)
    np.testing.assert_array_equal(ids_sq, ids_l2)
    np.testing.assert_allclose(np.sqrt(d_sq.astype(np.float64)),
                               d_l2.astype(np.float64), rtol=1e-6)


def test_quant_rescale():
    """scoreFromQuantized: raw/mul^2 before sqrt (ivfflat/search.go:1062-1077)."""
    vecs = np.array([[2, 0], [0, 2]], dtype=np.float32)
    cents = np.array([[0, 0]], dtype=np.float32)
    idx = orc.IvfIndex(cents, vecs, np.zeros(2, dtype=np.int64))
    q = np.zeros((1, 2), dtype=np.float32)
    ids, d = idx.search(orc.METRIC_L2, q, probe=1, k=2, orig_l2=True,
                        quant_mul=2.0)
    # raw l2sq = 4 in "quantized" domain, /4 = 1, sqrt = 1
    np.testing.assert_allclose(d[0], [1.0, 1.0])


def test_topk_merge():
    """cpu_topk_merge_sharded semantics: skip -1, dense ascending, pad."""
    ids = np.array([[[1, 2, -1]], [[3, -1, -1]]], dtype=np.int64)
    dists = np.array([[[0.5, 2.0, np.finfo(np.float32).max]],
                      [[1.0, np.finfo(np.float32).max, np.finfo(np.float32).max]]],
                     dtype=np.float32)
    out_ids, out_dists = orc.topk_merge(ids, dists, 3)
    np.testing.assert_array_equal(out_ids[0], [1, 3, 2])
    np.testing.assert_allclose(out_dists[0], [0.5, 1.0, 2.0])


def test_half_narrowing_restatements():
    """f16/bf16 narrowing (float16.go RTNE semantics): numpy float16 IS
    IEEE RTNE; the bf16 restatement must round-to-nearest-even, tie to
    even, quiet NaNs, and round-trip every representable bf16."""
    from oracle.oracle import encode_decode_bf16, encode_decode_f16
    # every representable bf16 value is a fixed point
    bits = np.arange(0, 1 << 16, dtype=np.uint32)
    vals = (bits << 16).view(np.float32)
    finite = np.isfinite(vals)
    rt = encode_decode_bf16(vals[finite])
    np.testing.assert_array_equal(rt, vals[finite])
    # ties round to even mantissa: 1 + 2^-8 * 1.5 exactly between two bf16
    x = np.float32(1.0 + 1.5 / 256.0)
    got = encode_decode_bf16(np.array([x]))[0]
    assert got in (np.float32(1.0 + 1.0 / 128.0),)  # even neighbor
    # NaN stays NaN (quieted), infinities preserved
    out = encode_decode_bf16(np.array([np.nan, np.inf, -np.inf],
                                      dtype=np.float32))
    assert np.isnan(out[0]) and np.isposinf(out[1]) and np.isneginf(out[2])
    # monotone on finite inputs
    rng = np.random.Generator(np.random.PCG64(2))
    xs = np.sort(rng.standard_normal(4096).astype(np.float32) * 100)
    ys = encode_decode_bf16(xs)
    assert (np.diff(ys) >= 0).all()
    zs = encode_decode_f16(np.clip(xs, -60000, 60000))
    assert (np.diff(zs) >= 0).all()
