"""ORACLE — test infrastructure only (never imported by the product path).

Drivers for the REFERENCE's own membership filters (oracle/_ref/
libref_filters.so: cgo/bloom.c, cbitmap.c, sorted64.c, croaring.c + vendored
CRoaring 4.7.0, compiled from the sources where they lie) plus a numpy/python
restatement of each membership test. Used by tests/test_docfilter.py to pin
matrixone_amd's independent payload parsers (moann_filter.cpp) bit-exactly to
the reference behavior, and by tests/golden/make_golden.py to produce the
committed golden payload fixtures that travel to the GPU box (where
/root/reference does not exist).

Kinds / payload formats (cgo/usearchex.h:30-33):
  bloom   : bloomfilter_t flat bytes (cgo/bloom.h:36-42)
  croaring: roaring64 PORTABLE bytes (cgo/croaring.c:99-112)
  cbitmap : [base u64][nbits u64][words] (cgo/cbitmap.c:159-201)
  sorted64: [count u64][sorted u64s]     (cgo/sorted64.c:26-57)
"""

from __future__ import annotations

import ctypes as ct
import os

import numpy as np

_REF = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                    "_ref", "libref_filters.so")

KIND_BLOOM = 0
KIND_CROARING = 2
KIND_CBITMAP = 3
KIND_SORTED64 = 4

_lib = None


def ref_lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_REF):
            raise RuntimeError(
                f"{_REF} missing — build with `make -C oracle filters` "
                "(requires /root/reference)")
        L = ct.CDLL(_REF)
        L.bloomfilter_init_with_seed.restype = ct.c_void_p
        L.bloomfilter_init_with_seed.argtypes = [ct.c_uint64, ct.c_uint32,
                                                 ct.c_uint64]
        L.bloomfilter_add.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_size_t]
        L.bloomfilter_test.restype = ct.c_bool
        L.bloomfilter_test.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_size_t]
        L.bloomfilter_free.argtypes = [ct.c_void_p]
        L.mo_cbitmap_build_fixed.restype = ct.c_int
        L.mo_cbitmap_build_fixed.argtypes = [
            ct.c_void_p, ct.c_size_t, ct.c_size_t, ct.c_size_t, ct.c_void_p,
            ct.c_size_t, ct.c_uint64, ct.c_int, ct.POINTER(ct.c_void_p)]
        L.mo_cbitmap_serialize.restype = ct.POINTER(ct.c_uint8)
        L.mo_cbitmap_serialize.argtypes = [ct.c_void_p,
                                           ct.POINTER(ct.c_size_t)]
        L.mo_cbitmap_contain.restype = ct.c_bool
        L.mo_cbitmap_contain.argtypes = [ct.c_void_p, ct.c_uint64]
        L.mo_croaring_create.restype = ct.c_void_p
        L.mo_croaring_add_fixed.restype = ct.c_bool
        L.mo_croaring_add_fixed.argtypes = [
            ct.c_void_p, ct.c_void_p, ct.c_size_t, ct.c_size_t, ct.c_size_t,
            ct.c_void_p, ct.c_size_t]
        L.mo_croaring_run_optimize.restype = ct.c_bool
        L.mo_croaring_run_optimize.argtypes = [ct.c_void_p]
        L.mo_croaring_contains.restype = ct.c_bool
        L.mo_croaring_contains.argtypes = [ct.c_void_p, ct.c_uint64]
        L.mo_croaring_serialize.restype = ct.POINTER(ct.c_uint8)
        L.mo_croaring_serialize.argtypes = [ct.c_void_p,
                                            ct.POINTER(ct.c_size_t)]
        L.mo_croaring_free.argtypes = [ct.c_void_p]
        L.mo_sorted64_contains.restype = ct.c_bool
        L.mo_sorted64_contains.argtypes = [ct.c_void_p, ct.c_uint64]
        _lib = L
    return _lib


# ---------------- builders (reference engine) -> serialized payloads --------

def build_bloom(keys: np.ndarray, nbits: int, k: int, seed: int) -> bytes:
    """bloomfilter_init_with_seed + add(key8) per key; payload = the flat
    bloomfilter_t bytes (header 24 B + bitmap). nbits is rounded up to a
    power of two by the reference (bloom.c:96-118)."""
    L = ref_lib()
    bf = L.bloomfilter_init_with_seed(nbits, k, seed)
    assert bf
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    for kk in keys:
        v = ct.c_int64(int(kk))
        L.bloomfilter_add(bf, ct.byref(v), 8)
    # read back the struct: magic(4) k(4) nbits(8) seed(8) bitmap
    hdr = ct.string_at(bf, 24)
    real_nbits = int.from_bytes(hdr[8:16], "little")
    nbytes = ((real_nbits + 63) // 64) * 8
    payload = ct.string_at(bf, 24 + nbytes)
    L.bloomfilter_free(bf)
    return payload


def ref_bloom_test(payload: bytes, keys: np.ndarray) -> np.ndarray:
    """Reference bloomfilter_test over a payload (reconstructed in place —
    the payload IS the struct layout)."""
    L = ref_lib()
    buf = ct.create_string_buffer(payload, len(payload))
    out = np.zeros(len(keys), dtype=bool)
    for i, kk in enumerate(np.ascontiguousarray(keys, dtype=np.int64)):
        v = ct.c_int64(int(kk))
        out[i] = L.bloomfilter_test(ct.cast(buf, ct.c_void_p),
                                    ct.byref(v), 8)
    return out


def build_cbitmap(keys: np.ndarray, use_offset: bool = True,
                  max_bits: int = 1 << 32) -> bytes:
    L = ref_lib()
    keys = np.ascontiguousarray(keys, dtype=np.uint64)
    out = ct.c_void_p()
    rc = L.mo_cbitmap_build_fixed(keys.ctypes.data, keys.nbytes, 8,
                                  len(keys), None, 0, max_bits,
                                  1 if use_offset else 0, ct.byref(out))
    assert rc == 0, rc
    ln = ct.c_size_t()
    buf = L.mo_cbitmap_serialize(out, ct.byref(ln))
    payload = ct.string_at(buf, ln.value)
    return payload


def build_croaring(keys: np.ndarray, run_optimize: bool = True) -> bytes:
    L = ref_lib()
    keys = np.ascontiguousarray(keys, dtype=np.uint64)
    r = L.mo_croaring_create()
    assert L.mo_croaring_add_fixed(r, keys.ctypes.data, keys.nbytes, 8,
                                   len(keys), None, 0)
    if run_optimize:
        L.mo_croaring_run_optimize(r)
    ln = ct.c_size_t()
    buf = L.mo_croaring_serialize(r, ct.byref(ln))
    payload = ct.string_at(buf, ln.value)
    L.mo_croaring_free(r)
    return payload


def build_sorted64(keys: np.ndarray) -> bytes:
    """[count u64][sorted u64s] — the Go side serializes exactly this."""
    vals = np.unique(np.ascontiguousarray(keys, dtype=np.uint64))
    return (np.uint64(len(vals)).tobytes() + vals.tobytes())


def ref_contains(kind: int, payload: bytes, keys: np.ndarray) -> np.ndarray:
    """Membership via the reference's own test functions on the payload."""
    L = ref_lib()
    keys64 = np.ascontiguousarray(keys, dtype=np.int64)
    out = np.zeros(len(keys64), dtype=bool)
    buf = ct.create_string_buffer(payload, len(payload))
    p = ct.cast(buf, ct.c_void_p)
    if kind == KIND_BLOOM:
        return ref_bloom_test(payload, keys64)
    if kind == KIND_SORTED64:
        for i, kk in enumerate(keys64):
            out[i] = L.mo_sorted64_contains(p, ct.c_uint64(int(kk) & (2**64 - 1)))
        return out
    if kind == KIND_CBITMAP:
        # reconstruct the live handle through the reference deserializer
        L.mo_cbitmap_deserialize.restype = ct.c_void_p
        L.mo_cbitmap_deserialize.argtypes = [ct.c_void_p, ct.c_size_t]
        h = L.mo_cbitmap_deserialize(p, len(payload))
        assert h
        for i, kk in enumerate(keys64):
            out[i] = L.mo_cbitmap_contain(h, ct.c_uint64(int(kk) & (2**64 - 1)))
        return out
    if kind == KIND_CROARING:
        L.mo_croaring_deserialize.restype = ct.c_void_p
        L.mo_croaring_deserialize.argtypes = [ct.c_void_p, ct.c_size_t]
        h = L.mo_croaring_deserialize(p, len(payload))
        assert h
        for i, kk in enumerate(keys64):
            out[i] = L.mo_croaring_contains(h, ct.c_uint64(int(kk) & (2**64 - 1)))
        L.mo_croaring_free(h)
        return out
    raise ValueError(kind)


# ---------------- python restatements (spec-level) --------------------------

def py_bloom_test(payload: bytes, keys: np.ndarray) -> np.ndarray:
    """bloom.c:33-79,177 restated with python-xxhash: probe (h1 + i*h2) &
    (nbits-1) of XXH3_128bits_withSeed(int64le, seed)."""
    import xxhash
    k = int.from_bytes(payload[4:8], "little")
    nbits = int.from_bytes(payload[8:16], "little")
    seed = int.from_bytes(payload[16:24], "little")
    words = np.frombuffer(payload[24:], dtype=np.uint64)
    out = np.zeros(len(keys), dtype=bool)
    if nbits == 0:
        return out
    for i, kk in enumerate(np.ascontiguousarray(keys, dtype=np.int64)):
        d = int(kk).to_bytes(8, "little", signed=True)
        h = xxhash.xxh3_128(d, seed=seed).intdigest()
        h1 = h & (2**64 - 1)          # XXH128 low64
        h2 = h >> 64                  # XXH128 high64
        ok = True
        for j in range(k):
            pos = (h1 + j * h2) & (nbits - 1)
            if not (int(words[pos >> 6]) >> (pos & 63)) & 1:
                ok = False
                break
        out[i] = ok
    return out


def py_cbitmap_test(payload: bytes, keys: np.ndarray) -> np.ndarray:
    base = int.from_bytes(payload[0:8], "little")
    nbits = int.from_bytes(payload[8:16], "little")
    words = np.frombuffer(payload[16:], dtype=np.uint64)
    out = np.zeros(len(keys), dtype=bool)
    for i, kk in enumerate(np.ascontiguousarray(keys, dtype=np.uint64)):
        v = int(kk)
        if v < base or v - base >= nbits:
            continue
        b = v - base
        out[i] = bool((int(words[b >> 6]) >> (b & 63)) & 1)
    return out


def py_sorted64_test(payload: bytes, keys: np.ndarray) -> np.ndarray:
    count = int.from_bytes(payload[0:8], "little")
    vals = np.frombuffer(payload[8:8 + count * 8], dtype=np.uint64)
    keys = np.ascontiguousarray(keys, dtype=np.uint64)
    idx = np.searchsorted(vals, keys)
    ok = idx < len(vals)
    out = np.zeros(len(keys), dtype=bool)
    out[ok] = vals[idx[ok]] == keys[ok]
    return out


def py_set_test(member_set, keys: np.ndarray) -> np.ndarray:
    """Exact-set semantics (the croaring restatement: membership of the set
    the filter was built from — the FORMAT itself is pinned by the golden
    payload + the reference deserializer)."""
    return np.array([int(kk) & (2**64 - 1) in member_set for kk in keys],
                    dtype=bool)
