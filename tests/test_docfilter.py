"""CPU tests for docfilter membership evaluation (moann_filter.cpp): the
four serialized filter kinds of cgo/usearchex.h:30-33 parsed independently
and pinned bit-exactly to the reference's own filter code.

Two layers:
  - golden fixtures (tests/golden/docfilter.npz, built by the reference's
    compiled bloom.c/cbitmap.c/sorted64.c/croaring.c via oracle/_ref) —
    runs ANYWHERE, including the GPU box where /root/reference is absent;
  - live cross-checks against oracle/_ref and the python restatements —
    skipped when the reference build is unavailable.
"""

import os

import numpy as np
import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = os.path.join(HERE, "golden", "docfilter.npz")

from matrixone_amd import engine  # noqa: E402


def _golden():
    return np.load(GOLDEN)


@pytest.mark.parametrize("kind,name", [
    (engine.FILTER_BLOOM, "bloom"),
    (engine.FILTER_CBITMAP, "cbitmap"),
    (engine.FILTER_SORTED64, "sorted64"),
])
def test_golden_kinds(kind, name):
    z = _golden()
    payload = z[f"{name}_payload"].tobytes()
    got = engine.docfilter_eval(kind, payload, z["probe_keys"])
    np.testing.assert_array_equal(got, z[f"{name}_expect"], err_msg=name)


def test_golden_croaring():
    z = _golden()
    got = engine.docfilter_eval(engine.FILTER_CROARING,
                                z["croaring_payload"].tobytes(),
                                z["croaring_probe_keys"])
    np.testing.assert_array_equal(got, z["croaring_expect"])


def test_unknown_kind_fails_closed():
    with pytest.raises(engine.MoannError):
        engine.docfilter_eval(9, b"\x00" * 64, np.arange(10))


def test_bad_payloads_fail():
    with pytest.raises(engine.MoannError):
        engine.docfilter_eval(engine.FILTER_BLOOM, b"NOPE" + b"\x00" * 32,
                              np.arange(4))
    with pytest.raises(engine.MoannError):
        engine.docfilter_eval(engine.FILTER_CROARING, b"\xff" * 40,
                              np.arange(4))
    # truncated exact filters match nothing rather than reading OOB
    assert not engine.docfilter_eval(engine.FILTER_CBITMAP, b"\x01",
                                     np.arange(4)).any()
    assert not engine.docfilter_eval(engine.FILTER_SORTED64, b"\x01",
                                     np.arange(4)).any()


_REF = os.path.join(HERE, "..", "oracle", "_ref", "libref_filters.so")
needs_ref = pytest.mark.skipif(not os.path.exists(_REF),
                               reason="reference filter build unavailable")


@needs_ref
@pytest.mark.parametrize("seed", [0, 1, 0xDEADBEEF, 2**63 + 11])
def test_bloom_vs_reference_live(seed):
    from oracle import docfilter as DF
    rng = np.random.Generator(np.random.PCG64(seed & 0xFFFF_FFFF))
    keys = rng.integers(-2**40, 2**40, 5000).astype(np.int64)
    probe = np.concatenate([keys[:300],
                            rng.integers(-2**40, 2**40, 4000)]).astype(np.int64)
    pay = DF.build_bloom(keys, nbits=1 << 15, k=7, seed=seed)
    mine = engine.docfilter_eval(engine.FILTER_BLOOM, pay, probe)
    ref = DF.ref_bloom_test(pay, probe)
    py = DF.py_bloom_test(pay, probe)
    np.testing.assert_array_equal(mine, ref)
    np.testing.assert_array_equal(py, ref)
    member = np.isin(probe, keys)
    assert mine[member].all()  # no false negatives, ever


@needs_ref
def test_croaring_container_shapes_live():
    """array, bitset, run containers and multiple 32-bit buckets."""
    from oracle import docfilter as DF
    rng = np.random.Generator(np.random.PCG64(4))
    sets = {
        "array": rng.choice(100_000, 900, replace=False),
        "bitset": np.arange(0, 200_000, 2),
        "runs": np.concatenate([np.arange(5, 9000),
                                np.arange(40_000, 41_000)]),
        "multi-bucket": np.concatenate(
            [rng.choice(5000, 500, replace=False) + (np.int64(b) << 32)
             for b in (0, 1, 7, 123)]),
    }
    sets["no-runopt"] = np.arange(8000, 60000, 3)  # cookie 12346 + offsets
    for name, kk in sets.items():
        kk = kk.astype(np.int64)
        pay = DF.build_croaring(kk, run_optimize=(name != "no-runopt"))
        probe = np.concatenate([kk[::3], kk + 1,
                                rng.integers(0, 2**40, 3000)]).astype(np.int64)
        mine = engine.docfilter_eval(engine.FILTER_CROARING, pay, probe)
        ref = DF.ref_contains(DF.KIND_CROARING, pay, probe)
        np.testing.assert_array_equal(mine, ref, err_msg=name)


@needs_ref
def test_cbitmap_offsets_live():
    from oracle import docfilter as DF
    rng = np.random.Generator(np.random.PCG64(6))
    # high base (offset layout) and legacy base-0
    for use_offset in (True, False):
        keys = (rng.choice(50_000, 3000, replace=False)
                + 10_000_000).astype(np.int64)
        pay = DF.build_cbitmap(keys, use_offset=use_offset)
        probe = np.concatenate([keys[:200], keys + 1,
                                [0, 9_999_999, 10_050_001]]).astype(np.int64)
        mine = engine.docfilter_eval(engine.FILTER_CBITMAP, pay, probe)
        ref = DF.ref_contains(DF.KIND_CBITMAP, pay, probe)
        np.testing.assert_array_equal(mine, ref)
        np.testing.assert_array_equal(DF.py_cbitmap_test(pay, probe), ref)


@needs_ref
def test_sorted64_live():
    from oracle import docfilter as DF
    rng = np.random.Generator(np.random.PCG64(8))
    keys = rng.choice(2**30, 4000, replace=False).astype(np.int64)
    pay = DF.build_sorted64(keys)
    probe = np.concatenate([keys[:500], keys + 1,
                            [0, 2**30 - 1]]).astype(np.int64)
    mine = engine.docfilter_eval(engine.FILTER_SORTED64, pay, probe)
    ref = DF.ref_contains(DF.KIND_SORTED64, pay, probe)
    np.testing.assert_array_equal(mine, ref)
    member = np.isin(probe, keys)
    np.testing.assert_array_equal(mine, member)


def test_empty_filters():
    # empty sorted64 / cbitmap match nothing
    assert not engine.docfilter_eval(engine.FILTER_SORTED64,
                                     np.uint64(0).tobytes(),
                                     np.arange(8)).any()
    pay = np.zeros(2, dtype=np.uint64).tobytes()  # base 0, nbits 0
    assert not engine.docfilter_eval(engine.FILTER_CBITMAP, pay,
                                     np.arange(8)).any()
