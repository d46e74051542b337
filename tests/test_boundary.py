"""CPU-only boundary tests: the C-ABI library loads, exports every symbol
include/moann.h declares, and fails loudly (no silent CPU fallback) when no
GPU is present."""

import ctypes as ct
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "moann.h")
LIB = os.path.join(REPO, "matrixone_amd", "libmoann_hip.so")


def _built():
    if not os.path.exists(LIB):
        import subprocess
        subprocess.run(["make", "-C",
                        os.path.join(REPO, "matrixone_amd", "csrc")],
                       check=True)
    return LIB


def _declared_functions():
    """Every function name declared in include/moann.h."""
    src = open(HEADER).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    src = re.sub(r"//.*", "", src)
    names = re.findall(r"\b(\w+)\s*\([^;{)]*(?:\([^)]*\))?[^;{]*\)\s*;", src)
    drop = {"defined"}
    return sorted(set(n for n in names if (n.startswith("gpu_ivf_flat_")
                                           or n.startswith("moann_"))
                  and n not in drop))


def test_header_declares_expected_surface():
    names = _declared_functions()
    # the reference-shaped core (cgo/cuvs/ivf_flat_c.h) must be present
    for required in ["gpu_ivf_flat_new_empty", "gpu_ivf_flat_add_chunk",
                     "gpu_ivf_flat_build", "gpu_ivf_flat_search",
                     "gpu_ivf_flat_search_async", "gpu_ivf_flat_search_wait",
                     "gpu_ivf_flat_get_neighbors", "gpu_ivf_flat_get_distances",
                     "gpu_ivf_flat_free_result", "gpu_ivf_flat_destroy",
                     "gpu_ivf_flat_cap", "gpu_ivf_flat_len",
                     "gpu_ivf_flat_get_n_list", "gpu_ivf_flat_get_centers",
                     "moann_brute_force_search", "moann_ivf_flat_perf"]:
        assert required in names, f"{required} missing from header"


def test_lib_exports_every_declared_symbol():
    L = ct.CDLL(_built())
    missing = [n for n in _declared_functions() if not hasattr(L, n)]
    assert not missing, f"libmoann_hip.so missing symbols: {missing}"


def test_version_and_device_count_work_without_gpu():
    from matrixone_amd import engine
    v = engine.lib().moann_version().decode()
    assert "gfx950" in v
    assert engine.device_count() >= 0


def test_fails_loudly_without_gpu():
    """Compute calls must error, never fall back to CPU."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present — fail-loud covered implicitly")
    import numpy as np
    from matrixone_amd import engine
    with pytest.raises(engine.MoannError):
        ix = engine.IvfFlatIndex(8, 2, capacity=4)
        ix.add(np.zeros((4, 8), dtype=np.float32))


def test_engine_rejects_bad_metric():
    from matrixone_amd import engine
    with pytest.raises(ValueError):
        engine.IvfFlatIndex(8, 2, metric="hamming")


def test_host_layer_semantics_no_gpu():
    """Host-side plumbing that must work GPU-free: metric maps, RuntimeConfig
    defaults, cache keying."""
    from matrixone_amd.ivfflat import (OP_TYPE_METRIC, IndexConfig,
                                       RuntimeConfig, DISTFN_L2)
    assert OP_TYPE_METRIC["vector_l2_ops"] == "l2sq"
    assert OP_TYPE_METRIC["vector_l2sq_ops"] == "l2sq"  # types.go:121-133
    rt = RuntimeConfig()
    assert rt.orig_func_name == DISTFN_L2
    cfg = IndexConfig(dimensions=128, lists=256)
    assert cfg.op_type == "vector_l2_ops"


def test_cache_load_once_and_remove():
    from matrixone_amd.cache import VectorIndexCache

    class FakeImpl:
        loads = 0

        def Load(self, proc):
            FakeImpl.loads += 1

        def Search(self, proc, query, rt):
            return [1], [0.5]

        def Destroy(self):
            pass

    c = VectorIndexCache(ttl=60)
    out1 = c.search("k1", FakeImpl, None, None, None)
    out2 = c.search("k1", FakeImpl, None, None, None)
    assert out1 == out2 == ([1], [0.5])
    assert FakeImpl.loads == 1  # load-once per entry (cache.go:376-403)
    c.remove("k1")
    c.search("k1", FakeImpl, None, None, None)
    assert FakeImpl.loads == 2  # reload after Remove
    c.destroy()


def test_batch_window_abi_present():
    """gpu_ivf_flat_set_batch_window (ivf_flat_c.h:91) is exported and
    settable without a GPU."""
    import ctypes as ct
    from matrixone_amd import engine
    L = engine.lib()
    assert hasattr(L, "gpu_ivf_flat_set_batch_window")


REF_IVF_HDR = "/root/reference/cgo/cuvs/ivf_flat_c.h"


@pytest.mark.skipif(not os.path.exists(REF_IVF_HDR),
                    reason="reference tree not present (GPU box)")
def test_full_reference_ivf_flat_abi_resolves():
    """EVERY function the reference's cgo binding can link against
    (cgo/cuvs/ivf_flat_c.h:39-199) resolves in libmoann_hip.so — the
    drop-in condition: the Go side keeps its call sites."""
    src = open(REF_IVF_HDR).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    src = re.sub(r"//.*", "", src)
    names = sorted(set(re.findall(r"\b(gpu_ivf_flat_\w+)\s*\(", src)))
    assert len(names) >= 30, names  # sanity: the full surface, not a stub
    L = ct.CDLL(_built())
    missing = [n for n in names if not hasattr(L, n)]
    assert not missing, f"reference ABI entries missing: {missing}"


def test_usearchex_membership_entry_resolves():
    """cgo/usearchex.h:38-42 usearchex_filtered_search_with_membership is
    exported under its exact reference name (the Go bridge swaps only the
    index handle it passes), plus the docfilter handle pair behind it."""
    L = ct.CDLL(_built())
    for n in ("usearchex_filtered_search_with_membership",
              "moann_docfilter_parse", "moann_docfilter_test",
              "moann_docfilter_free", "moann_hnsw_set_ef"):
        assert hasattr(L, n), n


def test_docfilter_handle_matches_eval():
    """moann_docfilter_parse + test == the one-shot moann_docfilter_eval on
    the same serialized payload (sorted64 kind; pure host code)."""
    import numpy as np
    from matrixone_amd import engine
    members = np.array(sorted([3, 17, 99, 1 << 40]), dtype=np.uint64)
    payload = np.concatenate([np.array([len(members)], dtype=np.uint64),
                              members]).tobytes()
    keys = np.array([1, 3, 17, 50, 99, 1 << 40, -1], dtype=np.int64)
    ref = engine.docfilter_eval(4, payload, keys)
    L = engine.lib()
    err = engine._Err()
    fh = L.moann_docfilter_parse(4, payload, len(payload), err.ref)
    err.check("parse")
    assert fh
    out = np.zeros((len(keys) + 31) // 32, dtype=np.uint32)
    err2 = engine._Err()
    rc = L.moann_docfilter_test(fh, keys.ctypes.data, len(keys),
                                out.ctypes.data, err2.ref)
    err2.check("test")
    assert rc == 0
    got = [(out[i // 32] >> (i % 32)) & 1 for i in range(len(keys))]
    assert list(got) == list(ref.astype(int))
    L.moann_docfilter_free(fh)
    # malformed payload fails closed with an error, not a handle
    err3 = engine._Err()
    bad = L.moann_docfilter_parse(99, payload, len(payload), err3.ref)
    assert not bad


@pytest.mark.gpu
def test_info_and_dynb_knob():
    """info() and the dynb knob on a fresh (unbuilt) index — host state
    only, but index construction touches hipSetDevice, hence gpu-marked."""
    from matrixone_amd import engine
    ix = engine.IvfFlatIndex(8, 4, metric="l2sq", capacity=0)
    inf = ix.info()
    assert inf["dimension"] == 8 and inf["n_lists"] == 4
    assert inf["status"] == "Empty" and inf["element_size"] == 4
    err = engine._Err()
    engine.lib().gpu_ivf_flat_set_dynb_conservative_dispatch(
        ix._h, True, err.ref)
    err.check("dynb")
    ix.close()


def test_traffic_and_mfma_parsers_pin_committed_evidence():
    """The committed PMC evidence is reproducible from the committed CSVs
    by the committed parsers (provenance chain for roofline.traffic and
    the MFMA ratio — CPU-only)."""
    import json
    import subprocess
    import sys as _sys
    import tempfile
    repo = REPO
    csvp = os.path.join(repo, "profiles", "r02_fetch_refine_qt16.csv")
    if not os.path.exists(csvp):
        pytest.skip("evidence csv not present")
    with tempfile.TemporaryDirectory() as td:
        outp = os.path.join(td, "t.json")
        r = subprocess.run(
            [_sys.executable, os.path.join(repo, "tools", "collect_traffic.py"),
             csvp, "--key", "k", "--kernel", "scan_i8_dot", "--out", outp],
            capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr
        got = json.load(open(outp))["k"]["bytes_per_launch"]
    want = json.load(open(os.path.join(repo, "profiles", "r02_traffic.json")))
    want = want["10000000x768-nlist4096-nprobe32-batch1024-refine-f32"][
        "bytes_per_launch"]
    assert got == want, (got, want)

    busy = os.path.join(repo, "profiles", "r02_mfma_busy.csv")
    wave = os.path.join(repo, "profiles", "r02_mfma_wave.csv")
    if os.path.exists(busy) and os.path.exists(wave):
        import json as _json
        r = subprocess.run(
            [_sys.executable, os.path.join(repo, "tools", "mfma_util.py"),
             busy, wave, "rank_gemm"], capture_output=True, text=True,
            timeout=120)
        assert r.returncode == 0, r.stderr
        got = _json.loads(r.stdout)
        want = _json.load(open(os.path.join(repo, "profiles",
                                            "r02_mfma_util.json")))
        assert got["mfma_busy_over_wave_cycles"] == \
            want["mfma_busy_over_wave_cycles"]


def test_load_file_rejects_garbage_no_gpu(tmp_path):
    """gpu_ivf_flat_load_file fails loudly on missing/corrupt containers
    BEFORE touching the GPU (magic check precedes device work)."""
    from matrixone_amd import engine
    L = ct.CDLL(_built())
    L.gpu_ivf_flat_load_file.restype = ct.c_void_p
    bp = engine.BuildParams(4, True, 0.5)
    dev = (ct.c_int * 1)(0)
    err = engine._Err()
    h = L.gpu_ivf_flat_load_file(str(tmp_path / "missing").encode(), 8,
                                 0, bp, dev, 1, 0, 0, 0, 0, err.ref)
    assert not h
    with pytest.raises(engine.MoannError):
        err.check("load_file")
    bad = tmp_path / "bad.moann"
    bad.write_bytes(b"NOTAPACK" + b"\x00" * 64)
    err2 = engine._Err()
    h2 = L.gpu_ivf_flat_load_file(str(bad).encode(), 8, 0, bp, dev, 1, 0,
                                  0, 0, 0, err2.ref)
    assert not h2
    with pytest.raises(engine.MoannError):
        err2.check("load_file")


def test_load_file_fuzz_no_crash(tmp_path):
    """Randomized corrupt pack containers: every parse must fail with an
    error (NULL handle + errmsg), never crash or succeed (CPU-only: the
    parse precedes device work)."""
    import numpy as np
    from matrixone_amd import engine
    L = ct.CDLL(_built())
    L.gpu_ivf_flat_load_file.restype = ct.c_void_p
    bp = engine.BuildParams(4, True, 0.5)
    dev = (ct.c_int * 1)(0)
    rng = np.random.Generator(np.random.PCG64(5150))
    magic = b"MOANNPK1\n"
    for i in range(24):
        kind = i % 4
        if kind == 0:  # pure noise
            blob = rng.integers(0, 256, rng.integers(0, 256),
                                dtype=np.uint8).tobytes()
        elif kind == 1:  # right magic, noise body
            blob = magic + rng.integers(0, 256, rng.integers(0, 128),
                                        dtype=np.uint8).tobytes()
        elif kind == 2:  # magic + plausible header, truncated payload
            name = b"manifest.json"
            blob = (magic + len(name).to_bytes(4, "little") + name +
                    (1 << 30).to_bytes(8, "little") + b"x" * 16)
        else:  # magic + oversized name length
            blob = magic + (4096).to_bytes(4, "little") + b"y" * 8
        f = tmp_path / f"fz{i}.moann"
        f.write_bytes(blob)
        err = engine._Err()
        h = L.gpu_ivf_flat_load_file(str(f).encode(), 8, 0, bp, dev, 1, 0,
                                     0, 0, 0, err.ref)
        assert not h, f"fuzz case {i} unexpectedly parsed"
        with pytest.raises(engine.MoannError):
            err.check("fuzz")
