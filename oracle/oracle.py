"""ORACLE — test infrastructure ONLY.

ctypes wrapper over oracle/liboracle.so (the CPU restatement of MatrixOne's
IVF-Flat/brute-force ANN hot path — see moann_oracle.c header for reference
cites) plus a driver for oracle/_ref/libusearch_c.so (the reference's OWN
usearch 2.25.3 + SimSIMD 6.5.3 arithmetic, used to PIN the restatement).

May be imported only by tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg. The product path (matrixone_amd/) never imports this.
"""

from __future__ import annotations

import ctypes as ct
import os
import subprocess

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))

# Metric enum — pkg/vectorindex/metric (see moann_oracle.c)
METRIC_L2 = 0
METRIC_L2SQ = 1
METRIC_IP = 2
METRIC_COS = 3
METRIC_L1 = 4

_f32p = np.ctypeslib.ndpointer(dtype=np.float32, flags="C_CONTIGUOUS")
_i64p = np.ctypeslib.ndpointer(dtype=np.int64, flags="C_CONTIGUOUS")


def _build_if_needed() -> None:
    lib = os.path.join(_HERE, "liboracle.so")
    if not os.path.exists(lib):
        subprocess.run(["make", "-s", "liboracle.so"], cwd=_HERE, check=True)


_oracle_lib = None


def lib() -> ct.CDLL:
    global _oracle_lib
    if _oracle_lib is None:
        _build_if_needed()
        L = ct.CDLL(os.path.join(_HERE, "liboracle.so"))
        L.moann_oracle_l2sq_f32.restype = ct.c_float
        L.moann_oracle_l2sq_f32.argtypes = [_f32p, _f32p, ct.c_int64]
        L.moann_oracle_ip_f32.restype = ct.c_float
        L.moann_oracle_ip_f32.argtypes = [_f32p, _f32p, ct.c_int64]
        L.moann_oracle_cos_f32.restype = ct.c_float
        L.moann_oracle_cos_f32.argtypes = [_f32p, _f32p, ct.c_int64]
        L.moann_oracle_l1_f32.restype = ct.c_float
        L.moann_oracle_l1_f32.argtypes = [_f32p, _f32p, ct.c_int64]
        L.moann_oracle_bruteforce.restype = None
        L.moann_oracle_bruteforce.argtypes = [
            ct.c_int, ct.c_int, _f32p, ct.c_int64, ct.c_int64, ct.c_void_p,
            _f32p, ct.c_int64, ct.c_int64, _i64p, _f32p,
        ]
        L.moann_oracle_ivf_search.restype = None
        L.moann_oracle_ivf_search.argtypes = [
            ct.c_int, ct.c_int, ct.c_double,
            _f32p, ct.c_int64, ct.c_int64,
            _f32p, ct.c_void_p, _i64p, _i64p,
            _f32p, ct.c_int64, ct.c_int64, ct.c_int64,
            _i64p, _f32p, ct.c_void_p,
        ]
        L.moann_oracle_topk_merge.restype = None
        L.moann_oracle_topk_merge.argtypes = [
            ct.c_int64, ct.c_int64, ct.c_int64, _i64p, _f32p, _i64p, _f32p,
        ]
        L.moann_oracle_num_threads.restype = ct.c_int
        _oracle_lib = L
    return _oracle_lib


def distance(metric: int, a: np.ndarray, b: np.ndarray) -> float:
    a = np.ascontiguousarray(a, dtype=np.float32)
    b = np.ascontiguousarray(b, dtype=np.float32)
    L = lib()
    fn = {
        METRIC_L2SQ: L.moann_oracle_l2sq_f32,
        METRIC_IP: L.moann_oracle_ip_f32,
        METRIC_COS: L.moann_oracle_cos_f32,
        METRIC_L1: L.moann_oracle_l1_f32,
    }
    if metric == METRIC_L2:
        return float(np.sqrt(L.moann_oracle_l2sq_f32(a, b, a.size)))
    return float(fn[metric](a, b, a.size))


def bruteforce(metric: int, base: np.ndarray, queries: np.ndarray, k: int,
               ids: np.ndarray | None = None, orig_l2: bool = False):
    """Exact ascending top-k (brute_force.go:336-509). Returns (ids, dists)."""
    base = np.ascontiguousarray(base, dtype=np.float32)
    queries = np.ascontiguousarray(queries, dtype=np.float32)
    nq = queries.shape[0]
    out_ids = np.empty((nq, k), dtype=np.int64)
    out_dists = np.empty((nq, k), dtype=np.float32)
    idp = None
    if ids is not None:
        ids = np.ascontiguousarray(ids, dtype=np.int64)
        idp = ids.ctypes.data_as(ct.c_void_p)
    lib().moann_oracle_bruteforce(
        metric, int(orig_l2), base, base.shape[0], base.shape[1], idp,
        queries, nq, k, out_ids, out_dists)
    return out_ids, out_dists


class IvfIndex:
    """CSR IVF index shared by the oracle and (via the same arrays) the HIP
    product path, so parity never depends on duplicated assignment logic."""

    def __init__(self, centroids: np.ndarray, vecs: np.ndarray,
                 assign: np.ndarray, ids: np.ndarray | None = None):
        self.centroids = np.ascontiguousarray(centroids, dtype=np.float32)
        self.vecs = np.ascontiguousarray(vecs, dtype=np.float32)
        self.ids = None if ids is None else np.ascontiguousarray(ids, np.int64)
        nlist = self.centroids.shape[0]
        assign = np.asarray(assign, dtype=np.int64)
        order = np.argsort(assign, kind="stable")
        counts = np.bincount(assign, minlength=nlist)
        self.list_offsets = np.zeros(nlist + 1, dtype=np.int64)
        np.cumsum(counts, out=self.list_offsets[1:])
        self.slot_rows = np.ascontiguousarray(order, dtype=np.int64)
        self.assign = assign

    def search(self, metric: int, queries: np.ndarray, probe: int, k: int,
               orig_l2: bool = False, quant_mul: float = 1.0,
               want_probe: bool = False):
        queries = np.ascontiguousarray(queries, dtype=np.float32)
        nq = queries.shape[0]
        out_ids = np.empty((nq, k), dtype=np.int64)
        out_dists = np.empty((nq, k), dtype=np.float32)
        probe_eff = min(probe, self.centroids.shape[0])
        out_probe = np.empty((nq, probe_eff), dtype=np.int64) if want_probe else None
        idp = None
        if self.ids is not None:
            idp = self.ids.ctypes.data_as(ct.c_void_p)
        pp = out_probe.ctypes.data_as(ct.c_void_p) if want_probe else None
        lib().moann_oracle_ivf_search(
            metric, int(orig_l2), float(quant_mul),
            self.centroids, self.centroids.shape[0], self.centroids.shape[1],
            self.vecs, idp, self.list_offsets, self.slot_rows,
            queries, nq, probe_eff, k, out_ids, out_dists, pp)
        if want_probe:
            return out_ids, out_dists, out_probe
        return out_ids, out_dists


def topk_merge(shard_ids: np.ndarray, shard_dists: np.ndarray, k: int):
    """cpu_topk_merge_sharded restatement (helper.h:137-167).
    shard_ids/dists: [nshard, nq, k]."""
    shard_ids = np.ascontiguousarray(shard_ids, dtype=np.int64)
    shard_dists = np.ascontiguousarray(shard_dists, dtype=np.float32)
    nshard, nq, kk = shard_ids.shape
    assert kk == k
    out_ids = np.empty((nq, k), dtype=np.int64)
    out_dists = np.empty((nq, k), dtype=np.float32)
    lib().moann_oracle_topk_merge(nshard, nq, k, shard_ids, shard_dists,
                                  out_ids, out_dists)
    return out_ids, out_dists


# ---------------- reference arithmetic: usearch exact search ----------------

_USEARCH_SCALAR_F32 = 1  # usearch.h usearch_scalar_f32_k
_USEARCH_METRIC = {  # usearch.h usearch_metric_kind_t (cos=1, ip=2, l2sq=3)
    "cos": 1, "ip": 2, "l2sq": 3,
}


def usearch_lib() -> ct.CDLL | None:
    """The reference's own arithmetic (oracle/_ref/libusearch_c.so); None when
    not built and not buildable (no /root/reference)."""
    path = os.path.join(_HERE, "_ref", "libusearch_c.so")
    if not os.path.exists(path):
        if os.path.exists("/root/reference/thirdparties/usearch-2.25.3.tar.gz"):
            subprocess.run(["make", "-s", "usearch"], cwd=_HERE, check=True)
        if not os.path.exists(path):
            return None
    return ct.CDLL(path)


def usearch_exact_search(base: np.ndarray, queries: np.ndarray, k: int,
                         metric: str = "l2sq", threads: int = 0):
    """usearch_exact_search (usearch-2.25.3 c/usearch.h:486) — the reference
    exact-search arithmetic (SimSIMD kernels). Returns (keys u64, dists f32)."""
    L = usearch_lib()
    if L is None:
        raise RuntimeError("oracle/_ref/libusearch_c.so unavailable")
    base = np.ascontiguousarray(base, dtype=np.float32)
    queries = np.ascontiguousarray(queries, dtype=np.float32)
    nq, d = queries.shape
    n = base.shape[0]
    keys = np.empty((nq, k), dtype=np.uint64)
    dists = np.empty((nq, k), dtype=np.float32)
    err = ct.c_char_p(None)
    L.usearch_exact_search.restype = None
    L.usearch_exact_search(
        base.ctypes.data_as(ct.c_void_p), ct.c_size_t(n), ct.c_size_t(4 * d),
        queries.ctypes.data_as(ct.c_void_p), ct.c_size_t(nq), ct.c_size_t(4 * d),
        ct.c_int(_USEARCH_SCALAR_F32), ct.c_size_t(d),
        ct.c_int(_USEARCH_METRIC[metric]), ct.c_size_t(k), ct.c_size_t(threads),
        keys.ctypes.data_as(ct.c_void_p), ct.c_size_t(8 * k),
        dists.ctypes.data_as(ct.c_void_p), ct.c_size_t(4 * k),
        ct.byref(err))
    if err.value:
        raise RuntimeError(err.value.decode())
    return keys, dists


# ---------------- int8/uint8 scalar quantization (quantizer.go) -------------

def int8_params(vmin, vmax):
    """quantizer.go:90-100 Int8Params: mul=255/rng, add=-min*mul-128."""
    rng = vmax - vmin
    if not (rng > 0) or np.isinf(rng):
        return 1.0, 0.0
    mul = 255.0 / rng
    return mul, -vmin * mul + (-128.0)


def uint8_params(vmin, vmax):
    """quantizer.go:232-240 Uint8Params: add folds only the -min offset."""
    rng = vmax - vmin
    if not (rng > 0) or np.isinf(rng):
        return 1.0, 0.0
    mul = 255.0 / rng
    return mul, -vmin * mul


def _round_half_away(x):
    """math.Round semantics (ties away from zero) in float64."""
    return np.copysign(np.floor(np.abs(x) + 0.5), x)


def quantize(vecs_f32, mul, add, unsigned=False):
    """ApplyInt8/ApplyUint8 (quantizer.go:165-176): TWO separate float32
    roundings (product then sum — the entry SQL evaluates them as two f32
    ops), then round-half-away-from-zero + clamp (Float32ToInt8/Uint8,
    container/types/float16.go:227-239). NaN -> 0."""
    v = np.asarray(vecs_f32, dtype=np.float32)
    if mul == 1.0 and add == 0.0:
        s = v
    else:
        m = (v * np.float32(mul)).astype(np.float32)
        s = (m + np.float32(add)).astype(np.float32)
    r = _round_half_away(s.astype(np.float64))
    r = np.where(np.isnan(s), 0.0, r)
    if unsigned:
        return np.clip(r, 0, 255).astype(np.uint8)
    return np.clip(r, -128, 127).astype(np.int8)


def _int_metric_dists(metric, rows_i, q_i):
    """Integer-domain distances (distance_func_narrow.go:330-450): exact
    int64 accumulation, MO conventions."""
    a = rows_i.astype(np.int64)
    b = q_i.astype(np.int64)
    if metric in (METRIC_L2, METRIC_L2SQ):
        d = a - b
        return (d * d).sum(1).astype(np.float64)
    if metric == METRIC_IP:
        return (-(a * b).sum(1)).astype(np.float64)
    if metric == METRIC_L1:
        return np.abs(a - b).sum(1).astype(np.float64)
    dot = (a * b).sum(1).astype(np.float64)
    na2 = (a * a).sum(1).astype(np.float64)
    nb2 = float((b * b).sum())
    denom = np.sqrt(na2) * np.sqrt(nb2)
    sim = np.where(denom > 0, dot / np.maximum(denom, 1e-300), 0.0)
    sim = np.clip(sim, -1.0, 1.0)
    return np.where(denom > 0, 1.0 - sim, 1.0)


def ivf_search_quantized(idx: "IvfIndex", metric, queries, probe, k,
                         mul, add, unsigned=False, orig_l2=False):
    """Quantized-entry IVF probe: centroid rank in f32 (LoadCentroids keeps
    f32 centroids), entries AND query quantized by the same affine map,
    integer-domain distance, then scoreFromQuantized (raw/mul^2 for every
    metric — ivfflat/search.go:1062-1077) + the sqrt transform."""
    queries = np.ascontiguousarray(queries, dtype=np.float32)
    nq = queries.shape[0]
    # f32 centroid ranking via the C oracle (identical to the f32 path)
    _, _, probe_ids = idx.search(metric, queries, probe, 1, want_probe=True)
    codes = quantize(idx.vecs, mul, add, unsigned)
    out_ids = np.full((nq, k), -1, dtype=np.int64)
    out_d = np.full((nq, k), np.float32(np.finfo(np.float32).max))
    inv_mul2 = 1.0 / (mul * mul) if mul not in (0.0, 1.0) else 1.0
    for qi in range(nq):
        q_i = quantize(queries[qi], mul, add, unsigned)
        cand_rows, cand_d = [], []
        for l in probe_ids[qi]:
            if l < 0:
                continue
            lo, hi = idx.list_offsets[l], idx.list_offsets[l + 1]
            rows = idx.slot_rows[lo:hi]
            if len(rows) == 0:
                continue
            cand_rows.append(rows)
            cand_d.append(_int_metric_dists(metric, codes[rows], q_i))
        if not cand_rows:
            continue
        rows = np.concatenate(cand_rows)
        d = np.concatenate(cand_d)
        order = np.lexsort((rows, d))[:k]
        sel = rows[order]
        raw = d[order] * inv_mul2
        if orig_l2 and metric in (METRIC_L2, METRIC_L2SQ):
            raw = np.sqrt(raw)
        out_ids[qi, :len(order)] = (idx.ids[sel] if idx.ids is not None
                                    else sel)
        out_d[qi, :len(order)] = raw.astype(np.float32)
    return out_ids, out_d


# ---------------- half-precision (f16 / bf16) narrowing ---------------------
# MO's float16/bf16 quantization is a PLAIN CAST (quantizer.go:50-58): entries
# and the query are narrowed round-to-nearest-even and distances decode back
# to f32 (distance_func_narrow.go:27-52). numpy's float16 conversion IS IEEE
# RTNE (== float16.go f32bitsToF16bits); bf16 is restated bit-exactly from
# BF16FromFloat32 (float16.go:53-63, incl. the quiet-NaN rule).

def encode_decode_f16(x: np.ndarray) -> np.ndarray:
    """f32 -> binary16 (RTNE) -> f32."""
    with np.errstate(over="ignore"):  # overflow -> inf is the IEEE (and
        # f32bitsToF16bits:131-134) behavior, not an error
        return (np.asarray(x, dtype=np.float32).astype(np.float16)
                .astype(np.float32))


def encode_decode_bf16(x: np.ndarray) -> np.ndarray:
    """f32 -> bfloat16 (RTNE, NaN quieted) -> f32, bit-exact to
    types.BF16FromFloat32/ToFloat32."""
    x = np.ascontiguousarray(x, dtype=np.float32)
    u = x.view(np.uint32)
    nan = ((u >> 23) & 0xFF == 0xFF) & (u & 0x7FFFFF != 0)
    rounded = (u + np.uint32(0x7FFF) + ((u >> 16) & 1)) >> 16
    b16 = np.where(nan, (u >> 16) | np.uint32(0x0040), rounded)
    return ((b16.astype(np.uint32) << 16)).view(np.float32).reshape(x.shape)


def ivf_search_half(idx: "IvfIndex", metric: int, queries: np.ndarray,
                    probe: int, k: int, bf16: bool, orig_l2: bool = False):
    """Half-storage search spec: run the exact IVF oracle over the DECODED
    entries/centroids-as-stored and the DECODED query (entries narrowed at
    build, query narrowed at search — relation_search.go:56-63; distance =
    f32 arithmetic on decoded values)."""
    enc = encode_decode_bf16 if bf16 else encode_decode_f16
    dec_vecs = enc(idx.vecs)
    didx = IvfIndex(idx.centroids, dec_vecs, idx.assign, ids=idx.ids)
    return didx.search(metric, enc(queries), probe, k, orig_l2=orig_l2)
