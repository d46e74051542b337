"""ctypes binding over libmoann_hip.so — the C-ABI boundary of the MI355X
backend (include/moann.h, mirroring cgo/cuvs/ivf_flat_c.h).

FAIL-LOUD CONTRACT: this module raises immediately if the HIP extension is
missing or a call fails. There is NO CPU fallback anywhere in the product
path — the oracle under oracle/ is test infrastructure only and is never
imported here.
"""

from __future__ import annotations

import ctypes as ct
import os

import numpy as np

_PKG = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_PKG, "libmoann_hip.so")

# distance_type_t (include/moann.h, values from cgo/cuvs/cuvs_types.h)
METRICS = {
    "l2sq": 0,   # DistanceType_L2Expanded
    "l2": 1,     # DistanceType_L2SqrtExpanded (sqrt applied at gather)
    "cos": 2,    # DistanceType_CosineExpanded
    "l1": 3,     # DistanceType_L1
    "ip": 6,     # DistanceType_InnerProduct (MO convention -a.b)
}


class BuildParams(ct.Structure):
    _fields_ = [("n_lists", ct.c_uint32), ("add_data_on_build", ct.c_bool),
                ("kmeans_trainset_fraction", ct.c_double)]


class SearchParams(ct.Structure):
    _fields_ = [("n_probes", ct.c_uint32)]


class SearchRes(ct.Structure):
    _fields_ = [("result_ptr", ct.c_void_p)]


class PerfT(ct.Structure):
    _fields_ = [
        ("scan_launches", ct.c_uint64), ("scan_ms", ct.c_double),
        ("scan_bytes", ct.c_uint64), ("scan_rows", ct.c_uint64),
        ("rank_launches", ct.c_uint64), ("rank_ms", ct.c_double),
        ("rank_flops", ct.c_uint64),
        ("select_ms", ct.c_double), ("other_ms", ct.c_double),
    ]


_lib = None


def lib() -> ct.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        raise RuntimeError(
            f"moann HIP extension missing at {_LIB_PATH} — build it with "
            "`make -C matrixone_amd/csrc` or __graft_entry__.build(). "
            "There is no CPU fallback.")
    L = ct.CDLL(_LIB_PATH)
    L.gpu_ivf_flat_new_empty.restype = ct.c_void_p
    L.gpu_ivf_flat_new_empty.argtypes = [
        ct.c_uint64, ct.c_uint32, ct.c_int, BuildParams,
        ct.POINTER(ct.c_int), ct.c_int, ct.c_uint32, ct.c_int, ct.c_int,
        ct.c_int, ct.c_void_p, ct.c_void_p]
    L.gpu_ivf_flat_new.restype = ct.c_void_p
    L.gpu_ivf_flat_new.argtypes = [
        ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_int, BuildParams,
        ct.POINTER(ct.c_int), ct.c_int, ct.c_uint32, ct.c_int, ct.c_int,
        ct.c_int, ct.c_void_p, ct.c_void_p]
    L.gpu_ivf_flat_load_file.restype = ct.c_void_p
    L.gpu_ivf_flat_load_file.argtypes = [
        ct.c_char_p, ct.c_uint32, ct.c_int, BuildParams,
        ct.POINTER(ct.c_int), ct.c_int, ct.c_uint32, ct.c_int, ct.c_int,
        ct.c_int, ct.c_void_p]
    for f in (L.gpu_ivf_flat_add_chunk, L.gpu_ivf_flat_add_chunk_float,
              L.gpu_ivf_flat_add_chunk_quantize, L.gpu_ivf_flat_extend,
              L.gpu_ivf_flat_extend_float):
        f.argtypes = [
            ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_void_p, ct.c_void_p]
    L.gpu_ivf_flat_delete_id.argtypes = [ct.c_void_p, ct.c_int64, ct.c_void_p]
    L.gpu_ivf_flat_train_quantizer.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_void_p]
    L.gpu_ivf_flat_info.restype = ct.c_void_p  # char*, caller frees
    L.gpu_ivf_flat_info.argtypes = [ct.c_void_p, ct.c_void_p]
    L.gpu_ivf_flat_save.argtypes = [ct.c_void_p, ct.c_char_p, ct.c_void_p]
    L.gpu_ivf_flat_set_dynb_conservative_dispatch.argtypes = [
        ct.c_void_p, ct.c_bool, ct.c_void_p]
    L.gpu_ivf_flat_set_filter_columns.argtypes = [
        ct.c_void_p, ct.c_char_p, ct.c_uint64, ct.c_void_p]
    L.gpu_ivf_flat_add_filter_chunk.argtypes = [
        ct.c_void_p, ct.c_uint32, ct.c_void_p, ct.c_void_p, ct.c_uint64,
        ct.c_void_p]
    for f in (L.gpu_ivf_flat_search_with_filter,
              L.gpu_ivf_flat_search_quantize_with_filter):
        f.restype = SearchRes
        f.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32,
                      ct.c_uint32, SearchParams, ct.c_char_p, ct.c_void_p]
    L.gpu_ivf_flat_search_quantize_with_filter_async.restype = ct.c_uint64
    L.gpu_ivf_flat_search_quantize_with_filter_async.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_uint32,
        SearchParams, ct.c_char_p, ct.c_void_p]
    L.moann_docfilter_parse.restype = ct.c_void_p
    L.moann_docfilter_parse.argtypes = [ct.c_int, ct.c_void_p, ct.c_uint64,
                                        ct.c_void_p]
    L.moann_docfilter_test.restype = ct.c_int
    L.moann_docfilter_test.argtypes = [ct.c_void_p, ct.c_void_p,
                                       ct.c_uint64, ct.c_void_p, ct.c_void_p]
    L.moann_docfilter_free.argtypes = [ct.c_void_p]
    L.moann_hnsw_set_ef.argtypes = [ct.c_void_p, ct.c_uint32, ct.c_void_p]
    L.usearchex_filtered_search_with_membership.restype = ct.c_size_t
    L.usearchex_filtered_search_with_membership.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_int, ct.c_size_t, ct.c_void_p,
        ct.c_int, ct.c_void_p, ct.c_void_p, ct.POINTER(ct.c_char_p)]
    L.moann_ivf_flat_set_centroids.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint32, ct.c_void_p]
    L.moann_ivf_flat_set_assignments.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_void_p]
    for f in (L.gpu_ivf_flat_build, L.gpu_ivf_flat_start):
        f.argtypes = [ct.c_void_p, ct.c_void_p]
    L.gpu_ivf_flat_destroy.argtypes = [ct.c_void_p, ct.c_void_p]
    for f in (L.gpu_ivf_flat_search, L.gpu_ivf_flat_search_quantize,
              L.moann_ivf_flat_search_device):
        f.restype = SearchRes
        f.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32,
                      ct.c_uint32, SearchParams, ct.c_void_p]
    for f in (L.gpu_ivf_flat_search_async,
              L.gpu_ivf_flat_search_quantize_async):
        f.restype = ct.c_uint64
        f.argtypes = [
            ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_uint32,
            SearchParams, ct.c_void_p]
    L.gpu_ivf_flat_search_wait.restype = SearchRes
    L.gpu_ivf_flat_search_wait.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_void_p]
    L.gpu_ivf_flat_get_neighbors.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_void_p]
    L.gpu_ivf_flat_get_distances.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_void_p]
    L.gpu_ivf_flat_free_result.argtypes = [ct.c_void_p]
    for f in (L.gpu_ivf_flat_cap, L.gpu_ivf_flat_len):
        f.restype = ct.c_uint64
        f.argtypes = [ct.c_void_p]
    L.gpu_ivf_flat_get_n_list.restype = ct.c_uint32
    L.gpu_ivf_flat_get_n_list.argtypes = [ct.c_void_p]
    L.gpu_ivf_flat_set_batch_window.argtypes = [ct.c_void_p, ct.c_int64,
                                                ct.c_void_p]
    L.gpu_ivf_flat_set_quantizer.argtypes = [ct.c_void_p, ct.c_float,
                                             ct.c_float, ct.c_void_p]
    L.gpu_ivf_flat_get_quantizer.argtypes = [ct.c_void_p, ct.c_void_p,
                                             ct.c_void_p, ct.c_void_p]
    L.gpu_ivf_flat_save_dir.argtypes = [ct.c_void_p, ct.c_char_p, ct.c_void_p]
    L.gpu_ivf_flat_load_dir.argtypes = [ct.c_void_p, ct.c_char_p, ct.c_int,
                                        ct.c_void_p]
    L.moann_ivf_flat_search_filtered.restype = SearchRes
    L.moann_ivf_flat_search_filtered.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_uint32,
        SearchParams, ct.c_void_p, ct.c_uint64, ct.c_void_p]
    L.moann_ivf_flat_search_submit.restype = SearchRes
    L.moann_ivf_flat_search_submit.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_int, ct.c_uint64, ct.c_uint32,
        ct.c_uint32, SearchParams, ct.c_void_p]
    L.moann_ivf_flat_result_wait.restype = ct.c_int
    L.moann_ivf_flat_result_wait.argtypes = [ct.c_void_p, ct.c_void_p]
    L.moann_ivf_flat_enable_refine.argtypes = [ct.c_void_p, ct.c_uint32,
                                               ct.c_void_p]
    L.moann_docfilter_eval.restype = ct.c_int
    L.moann_docfilter_eval.argtypes = [
        ct.c_int, ct.c_void_p, ct.c_uint64, ct.c_void_p, ct.c_uint64,
        ct.c_void_p, ct.c_void_p]
    L.moann_ivf_flat_filter_from_docfilter.restype = ct.c_int
    L.moann_ivf_flat_filter_from_docfilter.argtypes = [
        ct.c_void_p, ct.c_int, ct.c_void_p, ct.c_uint64, ct.c_void_p,
        ct.c_void_p]
    L.moann_preds_eval.restype = ct.c_int
    L.moann_preds_eval.argtypes = [
        ct.c_char_p, ct.c_char_p, ct.POINTER(ct.c_void_p),
        ct.POINTER(ct.c_void_p), ct.c_void_p, ct.c_uint64, ct.c_void_p,
        ct.c_void_p]
    L.moann_ivf_flat_filter_from_preds.restype = ct.c_int
    L.moann_ivf_flat_filter_from_preds.argtypes = [
        ct.c_void_p, ct.c_char_p, ct.c_char_p, ct.POINTER(ct.c_void_p),
        ct.POINTER(ct.c_void_p), ct.c_void_p, ct.c_void_p]
    L.moann_ivf_flat_slots_of_ids.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_void_p, ct.c_void_p]
    L.moann_brute_force_search.argtypes = [
        ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_int, ct.c_void_p,
        ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_int, ct.c_void_p,
        ct.c_void_p, ct.c_void_p]
    L.moann_ivf_pq_new.restype = ct.c_void_p
    L.moann_ivf_pq_new.argtypes = [ct.c_uint32, ct.c_uint32, ct.c_uint32,
                                   ct.c_uint64, ct.POINTER(ct.c_int),
                                   ct.c_int, ct.c_void_p]
    L.moann_ivf_pq_set_centroids.argtypes = [ct.c_void_p, ct.c_void_p,
                                             ct.c_uint32, ct.c_void_p]
    L.moann_ivf_pq_set_codebooks.argtypes = [ct.c_void_p, ct.c_void_p,
                                             ct.c_void_p]
    L.moann_ivf_pq_add_codes.argtypes = [ct.c_void_p, ct.c_void_p,
                                         ct.c_uint64, ct.c_void_p, ct.c_void_p]
    L.moann_ivf_pq_set_assignments.argtypes = [ct.c_void_p, ct.c_void_p,
                                               ct.c_uint64, ct.c_void_p]
    L.moann_ivf_pq_build.argtypes = [ct.c_void_p, ct.c_void_p]
    L.moann_ivf_pq_search.restype = SearchRes
    L.moann_ivf_pq_save_dir.argtypes = [ct.c_void_p, ct.c_char_p,
                                        ct.c_void_p]
    L.moann_ivf_pq_load_dir.argtypes = [ct.c_void_p, ct.c_char_p,
                                        ct.c_void_p]
    L.moann_ivf_pq_search_filtered.restype = SearchRes
    L.moann_ivf_pq_search_filtered.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_uint32,
        SearchParams, ct.c_void_p, ct.c_uint64, ct.c_void_p]
    L.moann_ivf_pq_search.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_uint64,
                                      ct.c_uint32, ct.c_uint32, SearchParams,
                                      ct.c_void_p]
    L.moann_ivf_pq_perf.argtypes = [ct.c_void_p, ct.POINTER(PerfT)]
    L.moann_ivf_pq_destroy.argtypes = [ct.c_void_p, ct.c_void_p]
    L.moann_ivf_flat_perf.argtypes = [ct.c_void_p, ct.POINTER(PerfT)]
    L.moann_ivf_flat_perf_reset.argtypes = [ct.c_void_p]
    L.moann_pairwise_distance.argtypes = [
        ct.c_int, ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32,
        ct.c_int, ct.c_void_p, ct.c_void_p]
    L.moann_device_count.restype = ct.c_int
    L.moann_version.restype = ct.c_char_p
    _lib = L
    return L


class MoannError(RuntimeError):
    pass


class _Err:
    """errmsg out-param holder (char** per cgo/cuvs/helper.h:59)."""

    def __init__(self):
        self.p = ct.c_char_p(None)

    @property
    def ref(self):
        return ct.byref(self.p)

    def check(self, where: str):
        if self.p.value:
            msg = self.p.value.decode()
            raise MoannError(f"{where}: {msg}")


def device_count() -> int:
    return lib().moann_device_count()


def _as_f32(a) -> np.ndarray:
    arr = np.ascontiguousarray(a, dtype=np.float32)
    return arr


COL_I32, COL_I64, COL_F32, COL_F64, COL_U64 = 0, 1, 2, 3, 4
_COL_DTYPES = {0: np.int32, 1: np.int64, 2: np.float32, 3: np.float64,
               4: np.uint64}


def _col_arrays(cols_meta, columns, nulls):
    """-> (cols_json bytes, void** data, void** nulls) keeping refs alive."""
    import json as _json
    cols_json = _json.dumps([{"name": n, "type": t}
                             for n, t in cols_meta]).encode()
    keep = []
    data_arr = (ct.c_void_p * max(1, len(columns)))()
    null_arr = (ct.c_void_p * max(1, len(columns)))()
    for i, ((name, t), col) in enumerate(zip(cols_meta, columns)):
        a = np.ascontiguousarray(col, dtype=_COL_DTYPES[t])
        keep.append(a)
        data_arr[i] = a.ctypes.data
        nl = None if nulls is None else nulls[i]
        if nl is not None:
            w = np.ascontiguousarray(nl, dtype=np.uint32)
            keep.append(w)
            null_arr[i] = w.ctypes.data
        else:
            null_arr[i] = None
    return cols_json, data_arr, null_arr, keep


def preds_eval(cols_meta, columns, preds_json: str, nrows: int, nulls=None,
               host_ids=None) -> np.ndarray:
    """Evaluate a planner predicate list (cuVS eval_filter_bitmap_cpu
    shape) over columnar arrays -> bool[nrows]. cols_meta: [(name, type)]
    with type in COL_*; columns: array per column; nulls: optional u32
    bitset words per column; host_ids: int64 array for the virtual PK
    column (col == -1)."""
    cols_json, data_arr, null_arr, keep = _col_arrays(cols_meta, columns,
                                                      nulls)
    out = np.zeros((nrows + 31) // 32, dtype=np.uint32)
    hid = None
    if host_ids is not None:
        hid = np.ascontiguousarray(host_ids, dtype=np.int64)
        keep.append(hid)
    err = _Err()
    rc = lib().moann_preds_eval(
        cols_json, preds_json.encode(), data_arr, null_arr,
        None if hid is None else hid.ctypes.data, nrows, out.ctypes.data,
        err.ref)
    err.check("preds_eval")
    if rc != 0:
        raise MoannError("preds_eval failed")
    idx = np.arange(nrows)
    return ((out[idx >> 5] >> (idx & 31)) & 1).astype(bool)


FILTER_BLOOM = 0
FILTER_CROARING = 2
FILTER_CBITMAP = 3
FILTER_SORTED64 = 4


def docfilter_eval(kind: int, payload: bytes, keys) -> np.ndarray:
    """Evaluate a serialized MatrixOne membership filter (docfilter tag
    `kind`, cgo/usearchex.h:30-33) over int64 keys -> bool array.
    Payload formats per include/moann.h."""
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    nw = (len(keys) + 31) // 32
    out = np.zeros(max(1, nw), dtype=np.uint32)
    buf = ct.create_string_buffer(bytes(payload), len(payload))
    err = _Err()
    rc = lib().moann_docfilter_eval(kind, ct.cast(buf, ct.c_void_p),
                                    len(payload), keys.ctypes.data,
                                    len(keys), out.ctypes.data, err.ref)
    err.check("docfilter_eval")
    if rc != 0:
        raise MoannError("docfilter_eval failed")
    idx = np.arange(len(keys))
    return ((out[idx >> 5] >> (idx & 31)) & 1).astype(bool)


class IvfFlatIndex:
    """IVF-Flat index on one MI355X.

    Mirrors the lifecycle the reference Go binding drives through the C ABI
    (pkg/cuvs/ivf_flat.go:35-1125: New / AddChunk / Build / Search /
    SearchAsync / SearchWait), with the MatrixOne-CPU-path ingestion shape
    (explicit centroids + entry assignments from the hidden tables)."""

    QTYPES = {"f32": 0, "f16": 1, "int8": 2, "uint8": 3, "bf16": 4}  # quantization_t

    def __init__(self, dim: int, nlist: int, metric: str = "l2sq",
                 capacity: int = 0, device: int = 0, qtype: str = "f32",
                 btype: str = "f32"):
        if metric not in METRICS:
            raise ValueError(f"metric {metric!r} not in {sorted(METRICS)}")
        self.dim, self.nlist, self.metric = dim, nlist, metric
        self.device = device
        self.btype = btype
        err = _Err()
        bp = BuildParams(nlist, True, 0.5)
        dev = (ct.c_int * 1)(device)
        self._h = lib().gpu_ivf_flat_new_empty(
            capacity, dim, METRICS[metric], bp, dev, 1, 0, 0,
            self.QTYPES[btype], self.QTYPES[qtype], None, err.ref)
        err.check("new_empty")
        if not self._h:
            raise MoannError("gpu_ivf_flat_new_empty returned NULL")

    def add(self, vecs, ids=None):
        """f32 rows -> gpu_ivf_flat_add_chunk_float (the storage-typed
        gpu_ivf_flat_add_chunk takes T-typed rows per the reference)."""
        vecs = _as_f32(vecs)
        idp = None
        if ids is not None:
            ids = np.ascontiguousarray(ids, dtype=np.int64)
            idp = ids.ctypes.data
        err = _Err()
        lib().gpu_ivf_flat_add_chunk_float(self._h, vecs.ctypes.data,
                                           vecs.shape[0], idp, err.ref)
        err.check("add_chunk_float")

    def extend(self, vecs, ids=None):
        """Post-build append (gpu_ivf_flat_extend_float)."""
        vecs = _as_f32(vecs)
        idp = None
        if ids is not None:
            ids = np.ascontiguousarray(ids, dtype=np.int64)
            idp = ids.ctypes.data
        err = _Err()
        lib().gpu_ivf_flat_extend_float(self._h, vecs.ctypes.data,
                                        vecs.shape[0], idp, err.ref)
        err.check("extend_float")

    def delete_id(self, id_: int):
        err = _Err()
        lib().gpu_ivf_flat_delete_id(self._h, int(id_), err.ref)
        err.check("delete_id")

    def train_quantizer(self, train):
        train = _as_f32(train)
        err = _Err()
        lib().gpu_ivf_flat_train_quantizer(self._h, train.ctypes.data,
                                           train.shape[0], err.ref)
        err.check("train_quantizer")

    def info(self) -> dict:
        import json as _json
        err = _Err()
        p = lib().gpu_ivf_flat_info(self._h, err.ref)
        err.check("info")
        if not p:
            raise MoannError("info returned NULL")
        try:
            s = ct.string_at(p).decode()
        finally:
            ct.CDLL(None).free(ct.c_void_p(p))
        return _json.loads(s)

    def save_file(self, path: str):
        err = _Err()
        lib().gpu_ivf_flat_save(self._h, path.encode(), err.ref)
        err.check("save")

    def set_filter_columns(self, col_meta_json: str, total_count: int):
        err = _Err()
        lib().gpu_ivf_flat_set_filter_columns(
            self._h, col_meta_json.encode(), total_count, err.ref)
        err.check("set_filter_columns")

    def add_filter_chunk(self, col_idx: int, data, null_bitmap=None):
        data = np.ascontiguousarray(data)
        nbp = None
        if null_bitmap is not None:
            null_bitmap = np.ascontiguousarray(null_bitmap, dtype=np.uint32)
            nbp = null_bitmap.ctypes.data
        err = _Err()
        lib().gpu_ivf_flat_add_filter_chunk(
            self._h, col_idx, data.ctypes.data, nbp, data.shape[0], err.ref)
        err.check("add_filter_chunk")

    def search_with_filter(self, queries, k: int, nprobe: int,
                           preds_json: str):
        queries = _as_f32(queries)
        nq = queries.shape[0]
        err = _Err()
        res = lib().gpu_ivf_flat_search_quantize_with_filter(
            self._h, queries.ctypes.data, nq, queries.shape[1], k,
            SearchParams(nprobe),
            preds_json.encode() if preds_json else None, err.ref)
        err.check("search_quantize_with_filter")
        return self._read_result(res, nq, k, "search_with_filter")

    def set_centroids(self, centroids):
        centroids = _as_f32(centroids)
        err = _Err()
        lib().moann_ivf_flat_set_centroids(self._h, centroids.ctypes.data,
                                           centroids.shape[0], err.ref)
        err.check("set_centroids")

    def set_assignments(self, assignments):
        assignments = np.ascontiguousarray(assignments, dtype=np.int32)
        err = _Err()
        lib().moann_ivf_flat_set_assignments(self._h, assignments.ctypes.data,
                                             assignments.shape[0], err.ref)
        err.check("set_assignments")

    def build(self):
        err = _Err()
        lib().gpu_ivf_flat_build(self._h, err.ref)
        err.check("build")
        lib().gpu_ivf_flat_start(self._h, err.ref)
        err.check("start")

    def __len__(self):
        return lib().gpu_ivf_flat_len(self._h)

    def _read_result(self, res, nq, k, where):
        if not res.result_ptr:
            raise MoannError(f"{where} returned NULL result")
        ids = np.empty((nq, k), dtype=np.int64)
        dists = np.empty((nq, k), dtype=np.float32)
        L = lib()
        L.gpu_ivf_flat_get_neighbors(res.result_ptr, nq * k, ids.ctypes.data)
        L.gpu_ivf_flat_get_distances(res.result_ptr, nq * k, dists.ctypes.data)
        L.gpu_ivf_flat_free_result(res.result_ptr)
        if self.metric == "ip":
            # The C ABI emits the reference contract: true +a.b for IP
            # (cgo/cuvs/index_base.hpp:355-375 transform_distance). This
            # mirror plays the Go side and restores MO's -a.b convention
            # (distance_func.go:174-207); ±FLT_MAX sentinels preserved.
            sent = np.float32(np.finfo(np.float32).max)
            m = (dists != sent) & (dists != -sent)
            dists[m] = -dists[m]
        return ids, dists

    def search(self, queries, k: int, nprobe: int):
        """f32 queries -> gpu_ivf_flat_search_quantize (base-typed; the
        storage-typed gpu_ivf_flat_search takes T-typed queries)."""
        queries = _as_f32(queries)
        nq = queries.shape[0]
        err = _Err()
        res = lib().gpu_ivf_flat_search_quantize(
            self._h, queries.ctypes.data, nq, queries.shape[1], k,
            SearchParams(nprobe), err.ref)
        err.check("search_quantize")
        return self._read_result(res, nq, k, "search")

    def set_quantizer(self, vmin: float, vmax: float):
        err = _Err()
        lib().gpu_ivf_flat_set_quantizer(self._h, vmin, vmax, err.ref)
        err.check("set_quantizer")

    def save_dir(self, path: str):
        err = _Err()
        lib().gpu_ivf_flat_save_dir(self._h, path.encode(), err.ref)
        err.check("save_dir")

    def load_dir(self, path: str):
        err = _Err()
        lib().gpu_ivf_flat_load_dir(self._h, path.encode(), 0, err.ref)
        err.check("load_dir")

    def slots_of_ids(self, ids):
        ids = np.ascontiguousarray(ids, dtype=np.int64)
        out = np.empty(ids.size, dtype=np.int64)
        err = _Err()
        lib().moann_ivf_flat_slots_of_ids(self._h, ids.ctypes.data, ids.size,
                                          out.ctypes.data, err.ref)
        err.check("slots_of_ids")
        return out

    def filter_bitset(self, allowed_ids):
        """Bitset over stored slots with bits set for allowed_ids — the
        host-side filter build (cuVS eval_filter_bitmap_cpu shape)."""
        n = len(self)
        words = np.zeros((n + 31) // 32, dtype=np.uint32)
        slots = self.slots_of_ids(allowed_ids)
        slots = slots[slots >= 0]
        np.bitwise_or.at(words, slots // 32,
                         (np.uint32(1) << (slots % 32).astype(np.uint32)))
        return words

    def search_filtered(self, queries, k: int, nprobe: int, bitset_words):
        """Membership-filtered search."""
        queries = _as_f32(queries)
        bitset_words = np.ascontiguousarray(bitset_words, dtype=np.uint32)
        nq = queries.shape[0]
        err = _Err()
        res = lib().moann_ivf_flat_search_filtered(
            self._h, queries.ctypes.data, nq, queries.shape[1], k,
            SearchParams(nprobe), bitset_words.ctypes.data,
            bitset_words.size * 32, err.ref)
        err.check("search_filtered")
        return self._read_result(res, nq, k, "search_filtered")

    def search_device(self, queries_tensor, k: int, nprobe: int):
        """queries_tensor: a contiguous f32 torch CUDA tensor [nq][dim]; the
        caller must have synchronised the producing stream."""
        nq, qdim = queries_tensor.shape
        err = _Err()
        res = lib().moann_ivf_flat_search_device(
            self._h, ct.c_void_p(queries_tensor.data_ptr()), nq, qdim, k,
            SearchParams(nprobe), err.ref)
        err.check("search_device")
        return self._read_result(res, nq, k, "search_device")

    def search_submit(self, queries_tensor, k: int, nprobe: int):
        """Pipelined submit (moann_ivf_flat_search_submit): centroid rank +
        host job build run now, the list scan is enqueued; a second submit
        overlaps with this one's scan. At most 2 in flight, collect FIFO
        with search_collect."""
        nq, qdim = queries_tensor.shape
        err = _Err()
        res = lib().moann_ivf_flat_search_submit(
            self._h, ct.c_void_p(queries_tensor.data_ptr()), 1, nq, qdim, k,
            SearchParams(nprobe), err.ref)
        err.check("search_submit")
        if not res.result_ptr:
            raise MoannError("search_submit returned NULL result")
        return (res.result_ptr, nq, k)

    def search_collect(self, ticket):
        """Wait for a search_submit ticket and return (ids, dists)."""
        ptr, nq, k = ticket
        err = _Err()
        rc = lib().moann_ivf_flat_result_wait(ptr, err.ref)
        err.check("result_wait")
        if rc != 0:
            raise MoannError("result_wait failed")

        class _R:
            result_ptr = ptr
        return self._read_result(_R, nq, k, "search_collect")

    def enable_refine(self, depth: int):
        """Two-stage scan (moann_ivf_flat_enable_refine): byte first pass +
        exact f32 re-rank of the top-`depth`. 0 disables."""
        err = _Err()
        lib().moann_ivf_flat_enable_refine(self._h, depth, err.ref)
        err.check("enable_refine")

    def preds_bitset(self, cols_meta, columns, preds_json: str,
                     nulls=None) -> np.ndarray:
        """Slot bitset from a planner predicate list over ADD-order column
        arrays (moann_ivf_flat_filter_from_preds) — feed search_filtered."""
        cols_json, data_arr, null_arr, keep = _col_arrays(cols_meta, columns,
                                                          nulls)
        n = len(self)
        out = np.zeros(max(1, (n + 31) // 32), dtype=np.uint32)
        err = _Err()
        rc = lib().moann_ivf_flat_filter_from_preds(
            self._h, cols_json, preds_json.encode(), data_arr, null_arr,
            out.ctypes.data, err.ref)
        err.check("preds_bitset")
        if rc != 0:
            raise MoannError("preds_bitset failed")
        return out

    def search_pred_filtered(self, queries, k: int, nprobe: int, cols_meta,
                             columns, preds_json: str, nulls=None):
        """Search gated by a SQL residual-predicate list (§8f3)."""
        return self.search_filtered(
            queries, k, nprobe,
            self.preds_bitset(cols_meta, columns, preds_json, nulls))

    def docfilter_bitset(self, kind: int, payload: bytes) -> np.ndarray:
        """Slot bitset from a serialized docfilter: membership of every
        entry's id, in global slot order — feed to search_filtered."""
        n = len(self)
        nw = (n + 31) // 32
        out = np.zeros(max(1, nw), dtype=np.uint32)
        buf = ct.create_string_buffer(bytes(payload), len(payload))
        err = _Err()
        rc = lib().moann_ivf_flat_filter_from_docfilter(
            self._h, kind, ct.cast(buf, ct.c_void_p), len(payload),
            out.ctypes.data, err.ref)
        err.check("docfilter_bitset")
        if rc != 0:
            raise MoannError("docfilter_bitset failed")
        return out

    def search_docfiltered(self, queries, k: int, nprobe: int, kind: int,
                           payload: bytes):
        """Filtered search gated by a serialized docfilter (the
        usearchex_filtered_search_with_membership shape, usearchex.c:51-70,
        evaluated host-side into the slot bitset the kernels consume)."""
        return self.search_filtered(queries, k, nprobe,
                                    self.docfilter_bitset(kind, payload))

    def set_batch_window(self, window_us: int):
        """ivf_flat_c.h:91 — fuse concurrent search_async calls arriving
        within window_us into one device batch (0 disables)."""
        err = _Err()
        lib().gpu_ivf_flat_set_batch_window(self._h, window_us, err.ref)
        err.check("set_batch_window")

    def search_async(self, queries, k: int, nprobe: int):
        queries = _as_f32(queries)
        self._async_shape = (queries.shape[0], k)
        err = _Err()
        job = lib().gpu_ivf_flat_search_quantize_async(
            self._h, queries.ctypes.data, queries.shape[0], queries.shape[1],
            k, SearchParams(nprobe), err.ref)
        err.check("search_quantize_async")
        return job

    def search_wait(self, job: int):
        err = _Err()
        res = lib().gpu_ivf_flat_search_wait(self._h, job, err.ref)
        err.check("search_wait")
        nq, k = self._async_shape
        return self._read_result(res, nq, k, "search_wait")

    def perf(self) -> dict:
        p = PerfT()
        lib().moann_ivf_flat_perf(self._h, ct.byref(p))
        return {f[0]: getattr(p, f[0]) for f in PerfT._fields_}

    def perf_reset(self):
        lib().moann_ivf_flat_perf_reset(self._h)

    def close(self):
        if getattr(self, "_h", None):
            err = _Err()
            lib().gpu_ivf_flat_destroy(self._h, err.ref)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def pairwise_distance(metric: str, query, rows, device: int = 0):
    """1xN pairwise distances (the SQL distance-builtin batch —
    metric.PairwiseDistanceLaunch counterpart, MO conventions)."""
    query = _as_f32(query).reshape(-1)
    rows = _as_f32(rows)
    out = np.empty(rows.shape[0], dtype=np.float32)
    err = _Err()
    lib().moann_pairwise_distance(METRICS[metric], query.ctypes.data,
                                  rows.ctypes.data, rows.shape[0],
                                  rows.shape[1], device, out.ctypes.data,
                                  err.ref)
    err.check("pairwise_distance")
    return out


def search_with_tail(index, queries, k: int, nprobe: int,
                     tail_vecs, tail_ids, metric: str = None):
    """CDC-tail overflow search (ivfpq/search_gpu.go:222-428 shape): the
    built index covers the snapshot; rows appended since (the CDC tail,
    not yet merged into the index) are brute-forced and the two top-k
    blocks are merged with cpu_topk_merge_sharded semantics (skip -1,
    dense ascending, (-1, FLT_MAX) padding)."""
    metric = metric or getattr(index, "metric", "l2sq")
    main_ids, main_d = index.search(queries, k, nprobe)
    if tail_vecs is None or len(tail_vecs) == 0:
        return main_ids, main_d
    t_ids, t_d = brute_force_search(tail_vecs, queries, min(k, len(tail_vecs)),
                                    metric=metric, ids=tail_ids)
    nq = main_ids.shape[0]
    kk = t_ids.shape[1]
    if kk < k:  # pad the tail block to k for a uniform merge
        pad_i = np.full((nq, k - kk), -1, dtype=np.int64)
        pad_d = np.full((nq, k - kk), np.finfo(np.float32).max,
                        dtype=np.float32)
        t_ids = np.concatenate([t_ids, pad_i], 1)
        t_d = np.concatenate([t_d, pad_d], 1)
    both_i = np.stack([main_ids, t_ids])
    both_d = np.stack([main_d.astype(np.float32), t_d.astype(np.float32)])
    dd = both_d.copy()
    dd[both_i < 0] = np.finfo(np.float32).max
    flat_i = both_i.transpose(1, 0, 2).reshape(nq, -1)
    flat_d = dd.transpose(1, 0, 2).reshape(nq, -1)
    order = np.argsort(flat_d, axis=1, kind="stable")[:, :k]
    out_d = np.take_along_axis(flat_d, order, 1)
    out_i = np.take_along_axis(flat_i, order, 1)
    out_i[out_d >= np.finfo(np.float32).max] = -1
    return out_i, out_d


def brute_force_search(dataset, queries, k: int, metric: str = "l2sq",
                       ids=None, device: int = 0):
    """Exact GPU top-k (pkg/vectorindex/brute_force counterpart)."""
    dataset = _as_f32(dataset)
    queries = _as_f32(queries)
    nq = queries.shape[0]
    out_ids = np.empty((nq, k), dtype=np.int64)
    out_dists = np.empty((nq, k), dtype=np.float32)
    idp = None
    if ids is not None:
        ids = np.ascontiguousarray(ids, dtype=np.int64)
        idp = ids.ctypes.data
    err = _Err()
    lib().moann_brute_force_search(
        dataset.ctypes.data, dataset.shape[0], dataset.shape[1],
        METRICS[metric], idp, queries.ctypes.data, nq, k, device,
        out_ids.ctypes.data, out_dists.ctypes.data, err.ref)
    err.check("brute_force_search")
    return out_ids, out_dists


class IvfPqIndex:
    """IVF-PQ index: harness-trained codebooks + codes,
    LUT-in-LDS ADC search on the GPU. L2 only this round."""

    def __init__(self, dim: int, nsub: int, nlist: int, capacity: int = 0,
                 device: int = 0):
        err = _Err()
        dev = (ct.c_int * 1)(device)
        self.dim, self.nsub, self.nlist = dim, nsub, nlist
        self._h = lib().moann_ivf_pq_new(dim, nsub, nlist, capacity, dev, 1,
                                         err.ref)
        err.check("pq_new")
        if not self._h:
            raise MoannError("moann_ivf_pq_new returned NULL")

    def set_centroids(self, centroids):
        centroids = _as_f32(centroids)
        err = _Err()
        lib().moann_ivf_pq_set_centroids(self._h, centroids.ctypes.data,
                                         centroids.shape[0], err.ref)
        err.check("pq_set_centroids")

    def set_codebooks(self, cb):
        cb = _as_f32(cb)
        err = _Err()
        lib().moann_ivf_pq_set_codebooks(self._h, cb.ctypes.data, err.ref)
        err.check("pq_set_codebooks")

    def add_codes(self, codes, ids=None):
        codes = np.ascontiguousarray(codes, dtype=np.uint8)
        idp = None
        if ids is not None:
            ids = np.ascontiguousarray(ids, dtype=np.int64)
            idp = ids.ctypes.data
        err = _Err()
        lib().moann_ivf_pq_add_codes(self._h, codes.ctypes.data,
                                     codes.shape[0], idp, err.ref)
        err.check("pq_add_codes")

    def set_assignments(self, assignments):
        assignments = np.ascontiguousarray(assignments, dtype=np.int32)
        err = _Err()
        lib().moann_ivf_pq_set_assignments(self._h, assignments.ctypes.data,
                                           assignments.shape[0], err.ref)
        err.check("pq_set_assignments")

    def build(self):
        err = _Err()
        lib().moann_ivf_pq_build(self._h, err.ref)
        err.check("pq_build")

    def search_filtered(self, queries, k: int, nprobe: int, bitset_words):
        """Membership-filtered ADC search (same slot-bitset contract as
        IvfFlatIndex.search_filtered)."""
        queries = _as_f32(queries)
        bits = np.ascontiguousarray(bitset_words, dtype=np.uint32)
        nq = queries.shape[0]
        err = _Err()
        res = lib().moann_ivf_pq_search_filtered(
            self._h, queries.ctypes.data, nq, queries.shape[1], k,
            SearchParams(nprobe), bits.ctypes.data, len(bits), err.ref)
        err.check("pq_search_filtered")
        if not res.result_ptr:
            raise MoannError("pq_search_filtered returned NULL result")
        ids = np.empty((nq, k), dtype=np.int64)
        dists = np.empty((nq, k), dtype=np.float32)
        L = lib()
        L.gpu_ivf_flat_get_neighbors(res.result_ptr, nq * k, ids.ctypes.data)
        L.gpu_ivf_flat_get_distances(res.result_ptr, nq * k,
                                     dists.ctypes.data)
        L.gpu_ivf_flat_free_result(res.result_ptr)
        return ids, dists

    def search(self, queries, k: int, nprobe: int):
        queries = _as_f32(queries)
        nq = queries.shape[0]
        err = _Err()
        res = lib().moann_ivf_pq_search(self._h, queries.ctypes.data, nq,
                                        queries.shape[1], k,
                                        SearchParams(nprobe), err.ref)
        err.check("pq_search")
        if not res.result_ptr:
            raise MoannError("pq_search returned NULL result")
        ids = np.empty((nq, k), dtype=np.int64)
        dists = np.empty((nq, k), dtype=np.float32)
        L = lib()
        L.gpu_ivf_flat_get_neighbors(res.result_ptr, nq * k, ids.ctypes.data)
        L.gpu_ivf_flat_get_distances(res.result_ptr, nq * k, dists.ctypes.data)
        L.gpu_ivf_flat_free_result(res.result_ptr)
        return ids, dists

    def save_dir(self, path: str):
        err = _Err()
        lib().moann_ivf_pq_save_dir(self._h, path.encode(), err.ref)
        err.check("pq_save_dir")

    def load_dir(self, path: str):
        err = _Err()
        lib().moann_ivf_pq_load_dir(self._h, path.encode(), err.ref)
        err.check("pq_load_dir")

    def perf(self) -> dict:
        p = PerfT()
        lib().moann_ivf_pq_perf(self._h, ct.byref(p))
        return {f[0]: getattr(p, f[0]) for f in PerfT._fields_}

    def close(self):
        if getattr(self, "_h", None):
            err = _Err()
            lib().moann_ivf_pq_destroy(self._h, err.ref)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
