"""HnswSearch — the MatrixOne-side host mirror of the HNSW search
implementation (pkg/vectorindex/hnsw), backed by the MI355X kernel instead of
cgo → usearch.

Mirrors pkg/vectorindex/hnsw/search.go semantics: HnswParam
{M, efConstruction, efSearch, op_type} (hnsw_search.go:113-254), per-model
search + merge (search.go:89-183; multi-model handling is the embedding
host's loop — this class serves one model), and the usearch→MO score
transform DistanceTransformHnsw (metric/types.go:234-243: IP = usearch − 1,
L2 = sqrt of l2sq when the query's function is l2_distance).

The GPU index ingests a usearch-exported graph (entry slot, levels,
adjacency, vectors by slot) — the shape oracle/hnsw.py's export produces.
"""

from __future__ import annotations

import ctypes as ct
from dataclasses import dataclass
from typing import Optional

import numpy as np

from . import engine

# usearch metric kinds accepted by the C layer (include/moann.h HNSW section)
USEARCH_METRICS = {"l2sq": 0, "ip": 1, "cos": 2}

# op_type -> usearch metric (OpTypeToUsearchMetric, metric/types.go:152-163)
OP_TYPE_USEARCH = {
    "vector_l2_ops": "l2sq",
    "vector_l2sq_ops": "l2sq",
    "vector_ip_ops": "ip",
    "vector_cosine_ops": "cos",
}


def _bind(L):
    if getattr(L, "_hnsw_bound", False):
        return L
    L.moann_hnsw_new.restype = ct.c_void_p
    L.moann_hnsw_new.argtypes = [ct.c_uint32, ct.c_int, ct.c_uint64,
                                 ct.c_int64, ct.c_int32, ct.POINTER(ct.c_int),
                                 ct.c_int, ct.c_void_p]
    L.moann_hnsw_set_vectors.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_void_p]
    L.moann_hnsw_set_keys.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_void_p]
    L.moann_hnsw_set_level0.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_void_p,
                                        ct.c_int64, ct.c_void_p]
    L.moann_hnsw_set_upper.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_int64,
                                       ct.c_void_p, ct.c_void_p, ct.c_int64,
                                       ct.c_void_p]
    L.moann_hnsw_build.argtypes = [ct.c_void_p, ct.c_void_p]
    L.moann_hnsw_search.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_uint64,
                                    ct.c_uint32, ct.c_uint32, ct.c_uint32,
                                    ct.c_void_p, ct.c_void_p, ct.c_void_p]
    L.moann_hnsw_search_device.argtypes = L.moann_hnsw_search.argtypes
    L.moann_hnsw_search_filtered.argtypes = [
        ct.c_void_p, ct.c_void_p, ct.c_uint64, ct.c_uint32, ct.c_uint32,
        ct.c_uint32, ct.c_void_p, ct.c_void_p, ct.c_void_p]
    L.moann_hnsw_perf.argtypes = [ct.c_void_p, ct.POINTER(engine.PerfT)]
    L.moann_hnsw_perf_reset.argtypes = [ct.c_void_p]
    L.moann_hnsw_destroy.argtypes = [ct.c_void_p, ct.c_void_p]
    L._hnsw_bound = True
    return L


@dataclass
class HnswGraphData:
    """A usearch-exported graph (the arrays oracle/hnsw.py exports, or the
    embedding host's equivalent ingestion of a model file)."""
    dim: int
    metric: str          # "l2sq" | "ip" | "cos" (usearch kind)
    entry_slot: int
    max_level: int
    levels: np.ndarray   # [n] int32
    keys: np.ndarray     # [n] uint64/int64 slot -> key (MO pk)
    vecs: np.ndarray     # [n][dim] f32 by slot
    adj: list            # per level: (offsets[n+1] int64, neighbors u32)


def load_usearch_index(data) -> "HnswGraphData":
    """Parse a usearch-2.x index_dense serialization (the file format
    MatrixOne persists per HNSW mini-model — hnsw/model.go chunks exactly
    these bytes into the model table) into HnswGraphData, natively.

    Layout (usearch index_dense.hpp save_to_stream:1143-1209 +
    index.hpp save_to_stream:3703-3748, studied not copied):
      [u32 rows][u32 bytes_per_vector][rows x bpv vector bytes]
      [64 B dense head: "usearch", 3 x u16 version, 4 x u8 kinds,
       u64 count_present, u64 count_deleted, u64 dimensions, u8 multi]
      [index header: 5 x u64 size/connectivity/connectivity_base/
       max_level/entry_slot]
      [size x i16 node levels]
      [per node: u64 key, i16 level, then level-0 block
       (u32 count + connectivity_base x u32 slots) and `level` blocks of
       (u32 count + connectivity x u32 slots)]
    Scalar kinds f32/f16/bf16 are decoded to f32 (the metric evaluates in
    f32 either way on our kernels). `data` is a path or bytes."""
    if isinstance(data, (str, bytes)) and isinstance(data, str):
        with open(data, "rb") as f:
            raw = np.frombuffer(f.read(), dtype=np.uint8)
    else:
        raw = np.frombuffer(data, dtype=np.uint8)
    off = 0
    rows, bpv = raw[off:off + 8].view(np.uint32)
    off += 8
    vec_bytes = raw[off:off + rows * bpv]
    off += int(rows) * int(bpv)
    head = raw[off:off + 64]
    off += 64
    if bytes(head[:7]) != b"usearch":
        raise ValueError("not a usearch index file")
    kind_metric = int(head[13])
    kind_scalar = int(head[14])
    dims = int(head[33:41].view(np.uint64)[0])
    metric = {ord("e"): "l2sq", ord("i"): "ip", ord("c"): "cos"}.get(
        kind_metric)
    if metric is None:
        raise ValueError(f"unsupported usearch metric kind {kind_metric}")
    count_deleted = int(head[25:33].view(np.uint64)[0])
    if count_deleted:
        # MO's sync.go rebuilds models rather than deleting in place; a
        # file with soft-deleted members would surface them as live here
        raise ValueError("usearch index holds deleted entries — rebuild/"
                         "compact the model before ingestion")
    hdr = raw[off:off + 40].view(np.uint64)
    off += 40
    size, conn, conn_base, max_level, entry_slot = (int(x) for x in hdr)
    if size != rows:
        raise ValueError("vector matrix and graph size disagree "
                         f"({rows} vs {size})")
    levels = raw[off:off + size * 2].view(np.int16).astype(np.int32)
    off += size * 2

    # vectors -> f32 [size][dims]
    if kind_scalar == 11:    # f32
        vecs = vec_bytes.view(np.float32).reshape(size, dims).copy()
    elif kind_scalar == 12:  # f16
        vecs = (vec_bytes.view(np.float16).reshape(size, dims)
                .astype(np.float32))
    elif kind_scalar == 4:   # bf16
        u = vec_bytes.view(np.uint16).reshape(size, dims)
        vecs = (u.astype(np.uint32) << 16).view(np.float32).copy()
    else:
        raise ValueError(f"unsupported usearch scalar kind {kind_scalar}")

    # node offsets from the levels alone (fixed per-level block sizes)
    base_block = 4 + 4 * conn_base
    lvl_block = 4 + 4 * conn
    node_bytes = 10 + base_block + levels.astype(np.int64) * lvl_block
    node_off = np.zeros(size + 1, dtype=np.int64)
    np.cumsum(node_bytes, out=node_off[1:])
    node_off += off
    if int(node_off[-1]) > raw.size:
        raise ValueError("truncated usearch index file")

    keys = np.empty(size, dtype=np.uint64)
    k8 = node_off[:-1, None] + np.arange(8)
    keys[:] = raw[k8].reshape(size, 8).copy().view(np.uint64).ravel()

    adj = []
    for lvl in range(max_level + 1):
        has = np.nonzero(levels >= lvl)[0]
        offs = np.zeros(size + 1, dtype=np.int64)
        if len(has) == 0:
            adj.append((offs, np.zeros(0, dtype=np.uint32)))
            continue
        block = base_block if lvl == 0 else lvl_block
        boff = node_off[has] + 10 + (0 if lvl == 0 else
                                     base_block + (lvl - 1) * lvl_block)
        blocks = raw[boff[:, None] + np.arange(block)].reshape(len(has),
                                                               block)
        w = blocks.copy().view(np.uint32).reshape(len(has), block // 4)
        counts = w[:, 0].astype(np.int64)
        cap = block // 4 - 1
        if (counts > cap).any():
            raise ValueError("corrupt neighbor count in usearch file")
        per = np.zeros(size, dtype=np.int64)
        per[has] = counts
        np.cumsum(per, out=offs[1:])
        mask = np.arange(cap)[None, :] < counts[:, None]
        nbrs = w[:, 1:][mask].astype(np.uint32)
        if nbrs.size and int(nbrs.max()) >= size:
            raise ValueError("corrupt neighbor slot in usearch file")
        adj.append((offs, nbrs))

    return HnswGraphData(dim=dims, metric=metric, entry_slot=entry_slot,
                         max_level=max_level, levels=levels, keys=keys,
                         vecs=vecs, adj=adj)


class HnswIndex:
    """One HNSW model on one MI355X (ingested usearch graph)."""

    def __init__(self, g: HnswGraphData, device: int = 0):
        L = _bind(engine.lib())
        self.dim = g.dim
        self.metric = g.metric
        n = g.vecs.shape[0]
        err = engine._Err()
        dev = (ct.c_int * 1)(device)
        self._h = L.moann_hnsw_new(g.dim, USEARCH_METRICS[g.metric], n,
                                   g.entry_slot, g.max_level, dev, 1, err.ref)
        err.check("hnsw_new")
        if not self._h:
            raise engine.MoannError("moann_hnsw_new returned NULL")
        vecs = np.ascontiguousarray(g.vecs, dtype=np.float32)
        L.moann_hnsw_set_vectors(self._h, vecs.ctypes.data, err.ref)
        err.check("set_vectors")
        keys = np.ascontiguousarray(g.keys.astype(np.int64))
        L.moann_hnsw_set_keys(self._h, keys.ctypes.data, err.ref)
        err.check("set_keys")
        self._gkeys = keys
        offs0, nbrs0 = g.adj[0]
        offs0 = np.ascontiguousarray(offs0, dtype=np.int64)
        nbrs0 = np.ascontiguousarray(nbrs0, dtype=np.uint32)
        L.moann_hnsw_set_level0(self._h, offs0.ctypes.data, nbrs0.ctypes.data,
                                nbrs0.size, err.ref)
        err.check("set_level0")
        if g.max_level > 0:
            u_mask = g.levels >= 1
            u_slots = np.nonzero(u_mask)[0]
            u_count = len(u_slots)
            u_idx = np.full(n, -1, dtype=np.int32)
            u_idx[u_slots] = np.arange(u_count, dtype=np.int32)
            up_offs = np.zeros((g.max_level, u_count + 1), dtype=np.int64)
            nbr_chunks = []
            base = 0
            for lvl in range(1, g.max_level + 1):
                offs, nbrs = g.adj[lvl]
                # offsets restricted to upper nodes, rebased into the concat
                cnt = (offs[u_slots + 1] - offs[u_slots]).astype(np.int64)
                up_offs[lvl - 1, 1:] = np.cumsum(cnt)
                up_offs[lvl - 1, :] += base
                for s in u_slots:
                    if offs[s + 1] > offs[s]:
                        nbr_chunks.append(nbrs[offs[s]:offs[s + 1]])
                base = up_offs[lvl - 1, -1]
            up_nbrs = (np.concatenate(nbr_chunks).astype(np.uint32)
                       if nbr_chunks else np.empty(0, dtype=np.uint32))
            up_offs32 = np.ascontiguousarray(up_offs.reshape(-1),
                                             dtype=np.int32)
            u_idx = np.ascontiguousarray(u_idx)
            up_nbrs = np.ascontiguousarray(up_nbrs)
            L.moann_hnsw_set_upper(self._h, u_idx.ctypes.data, u_count,
                                   up_offs32.ctypes.data, up_nbrs.ctypes.data,
                                   up_nbrs.size, err.ref)
            err.check("set_upper")
        L.moann_hnsw_build(self._h, err.ref)
        err.check("hnsw_build")

    def search(self, queries, ef: int, k: int):
        """Raw usearch-convention results (ids by key, dists)."""
        L = _bind(engine.lib())
        queries = np.ascontiguousarray(queries, dtype=np.float32)
        nq = queries.shape[0]
        out_ids = np.empty((nq, k), dtype=np.int64)
        out_dists = np.empty((nq, k), dtype=np.float32)
        err = engine._Err()
        L.moann_hnsw_search(self._h, queries.ctypes.data, nq,
                            queries.shape[1], ef, k, out_ids.ctypes.data,
                            out_dists.ctypes.data, err.ref)
        err.check("hnsw_search")
        return out_ids, out_dists

    @classmethod
    def from_usearch_file(cls, path_or_bytes):
        """Build the GPU index directly from a usearch-serialized
        mini-model (the bytes MO's hnsw/model.go persists)."""
        return cls(load_usearch_index(path_or_bytes))

    def search_docfiltered(self, queries, ef: int, k: int, kind: int,
                           payload: bytes):
        """Filtered beam search gated by a serialized docfilter — the
        usearchex_filtered_search_with_membership shape (usearchex.c:51-70):
        each graph key (MO pk) is tested against the filter host-side and
        the passing set becomes the slot bitset the kernel consumes."""
        bits = engine.docfilter_eval(kind, payload,
                                     self._gkeys.astype(np.int64))
        slots = np.nonzero(bits)[0]
        words = np.zeros((len(self._gkeys) + 31) // 32, dtype=np.uint32)
        np.bitwise_or.at(words, slots // 32,
                         (np.uint32(1) << (slots % 32).astype(np.uint32)))
        return self.search_filtered(queries, ef, k, words)

    def search_filtered(self, queries, ef: int, k: int,
                        slot_bitset: np.ndarray):
        """Membership-filtered search (usearchex semantics; §8a4)."""
        L = _bind(engine.lib())
        queries = np.ascontiguousarray(queries, dtype=np.float32)
        slot_bitset = np.ascontiguousarray(slot_bitset, dtype=np.uint32)
        nq = queries.shape[0]
        out_ids = np.empty((nq, k), dtype=np.int64)
        out_dists = np.empty((nq, k), dtype=np.float32)
        err = engine._Err()
        L.moann_hnsw_search_filtered(self._h, queries.ctypes.data, nq,
                                     queries.shape[1], ef, k,
                                     slot_bitset.ctypes.data,
                                     out_ids.ctypes.data,
                                     out_dists.ctypes.data, err.ref)
        err.check("hnsw_search_filtered")
        return out_ids, out_dists

    def search_device(self, queries_tensor, ef: int, k: int):
        L = _bind(engine.lib())
        nq, qdim = queries_tensor.shape
        out_ids = np.empty((nq, k), dtype=np.int64)
        out_dists = np.empty((nq, k), dtype=np.float32)
        err = engine._Err()
        L.moann_hnsw_search_device(self._h,
                                   ct.c_void_p(queries_tensor.data_ptr()),
                                   nq, qdim, ef, k, out_ids.ctypes.data,
                                   out_dists.ctypes.data, err.ref)
        err.check("hnsw_search_device")
        return out_ids, out_dists

    def perf(self) -> dict:
        p = engine.PerfT()
        _bind(engine.lib()).moann_hnsw_perf(self._h, ct.byref(p))
        return {f[0]: getattr(p, f[0]) for f in engine.PerfT._fields_}

    def perf_reset(self):
        _bind(engine.lib()).moann_hnsw_perf_reset(self._h)

    def close(self):
        if getattr(self, "_h", None):
            err = engine._Err()
            _bind(engine.lib()).moann_hnsw_destroy(self._h, err.ref)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class HnswSearch:
    """VectorIndexSearchIf mirror for HNSW (cache.go:65-84), one model.
    Applies DistanceTransformHnsw like hnsw/search.go:167-182."""

    def __init__(self, graph: HnswGraphData, op_type: str = "vector_l2_ops",
                 ef_search: int = 64, device: int = 0):
        self.graph = graph
        self.op_type = op_type
        self.ef_search = ef_search
        self.device = device
        self.index: Optional[HnswIndex] = None

    def Load(self, proc=None) -> None:
        self.index = HnswIndex(self.graph, device=self.device)

    def Search(self, proc, query, rt):
        if self.index is None:
            raise RuntimeError("index not loaded — call Load() first")
        q = np.atleast_2d(np.asarray(query, dtype=np.float32))
        ids, dists = self.index.search(q, self.ef_search, rt.limit)
        d64 = dists.astype(np.float64)
        valid = ids >= 0
        # DistanceTransformHnsw (metric/types.go:234-243)
        if self.graph.metric == "ip":
            d64 = np.where(valid, d64 - 1.0, d64)
        elif (self.graph.metric == "l2sq"
              and rt.orig_func_name == "l2_distance"):
            d64 = np.where(valid, np.sqrt(np.maximum(d64, 0.0)), d64)
        return ids, d64

    def SearchFloat32(self, proc, query, rt, out_keys, out_dists):
        ids, d64 = self.Search(proc, query, rt)
        n = ids.size
        out_keys[:n] = ids.reshape(-1)
        out_dists[:n] = d64.reshape(-1).astype(np.float32)

    def Destroy(self):
        if self.index is not None:
            self.index.close()
            self.index = None


class MultiModelHnswSearch:
    """Multi-model fan-out — the reference splits an HNSW index into
    mini-models of <= hnsw_max_index_capacity vectors (sync.go:98-106) and
    searches them concurrently, merging in a bounded SearchResultSafeHeap
    (hnsw/search.go:89-183, per-shard k clamp :189-201). Here: one GPU index
    per model, sequential device searches (one stream), dense merge with the
    cpu_topk_merge_sharded semantics."""

    def __init__(self, graphs, op_type: str = "vector_l2_ops",
                 ef_search: int = 64, device: int = 0):
        self.searchers = [HnswSearch(g, op_type, ef_search, device)
                          for g in graphs]

    def Load(self, proc=None):
        for sch in self.searchers:
            sch.Load()

    def Search(self, proc, query, rt):
        blocks_i, blocks_d = [], []
        for sch in self.searchers:
            ids, d64 = sch.Search(proc, query, rt)
            blocks_i.append(ids)
            blocks_d.append(d64)
        ids = np.stack(blocks_i)   # [models][nq][k]
        d = np.stack(blocks_d)
        nm, nq, k = ids.shape
        flat_i = ids.transpose(1, 0, 2).reshape(nq, nm * k)
        flat_d = d.transpose(1, 0, 2).reshape(nq, nm * k)
        fmax = np.finfo(np.float64).max
        flat_d = np.where(flat_i < 0, fmax, flat_d)
        order = np.lexsort((flat_i, flat_d), axis=1)[:, :k]
        out_d = np.take_along_axis(flat_d, order, 1)
        out_i = np.take_along_axis(flat_i, order, 1)
        out_i[out_d >= fmax] = -1
        out_d = np.where(out_i < 0, np.float64(np.finfo(np.float32).max), out_d)
        return out_i, out_d

    def Destroy(self):
        for sch in self.searchers:
            sch.Destroy()


class ShardedHnswSearch:
    """Multi-GPU HNSW model sharding (HNSW side): the
    reference already searches disjoint mini-models concurrently on one
    host (hnsw/search.go:140-160); here each RANK owns the models assigned
    round-robin to it (one process per GPU over torch.distributed) and one
    RCCL all-gather of the per-rank top-k blocks merges the answer —
    exactly the IVF exchange (matrixone_amd/distributed.py), since model
    results carry globally-unique keys.

    Construction: pass the FULL model list; rank r keeps models
    [r::world]. Ranks with no models still participate in the collective
    with sentinel blocks."""

    def __init__(self, graphs, op_type: str = "vector_l2_ops",
                 ef_search: int = 64, device: int = 0, rank: int = 0,
                 world: int = 1):
        self.rank, self.world = rank, world
        mine = list(graphs)[rank::world]
        self.local = (MultiModelHnswSearch(mine, op_type, ef_search, device)
                      if mine else None)

    def Load(self, proc=None):
        if self.local:
            self.local.Load()

    def Search(self, proc, query, rt):
        k = rt.limit
        q = np.atleast_2d(np.asarray(query, dtype=np.float32))
        nq = q.shape[0]
        if self.local:
            ids, d64 = self.local.Search(proc, q, rt)
        else:
            ids = np.full((nq, k), -1, dtype=np.int64)
            d64 = np.full((nq, k), np.float64(np.finfo(np.float32).max))
        if self.world == 1:
            return ids, d64
        from matrixone_amd.distributed import allgather_merge
        m_ids, m_d = allgather_merge(ids, d64.astype(np.float32), k)
        return m_ids, m_d.astype(np.float64)

    def Destroy(self):
        if self.local:
            self.local.Destroy()
