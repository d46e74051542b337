"""ORACLE — test infrastructure ONLY.

HNSW parity tooling: (a) ctypes driver for oracle/_ref/libref_hnsw.so — the
reference's own usearch 2.25.3 engine (build / search / graph export), and
(b) a numpy restatement of usearch's search algorithm over the exported
graph, pinned against (a). The GPU product kernel is parity-tested against
both on the SAME exported graph.

Algorithm restated from the vendored usearch-2.25.3 index.hpp:
  - search entry (index.hpp:3455-3498): greedy descent `search_for_one_`
    from entry_slot at max_level down to level 1, then base-level beam
    `search_to_find_in_base_` with expansion ef, then sort ascending and
    shrink to k.
  - search_for_one_ (index.hpp:4466-4506): per level, repeatedly scan the
    current node's neighbors in tape order and move to any strictly closer
    one until no improvement.
  - search_to_find_in_base_ (index.hpp:4688-4758): min-heap `next`, bounded
    sorted buffer `top` (capacity ef), visited set; push start into both;
    pop the closest candidate, stop when top is full and the candidate is
    farther than the current radius (top's max); neighbors are pushed to
    BOTH next and top iff (top not full or dist < radius).
usearch metric values (l2sq; ip = 1 - a.b; cos = 1 - cossim) — the MO-side
transform (DistanceTransformHnsw, metric/types.go:234-243) happens above.
"""

from __future__ import annotations

import ctypes as ct
import heapq
import os
from dataclasses import dataclass

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))

METRIC_L2SQ = 0
METRIC_IP = 1   # usearch ip: 1 - a.b
METRIC_COS = 2  # usearch cos: 1 - cossim

_lib = None


def lib():
    global _lib
    if _lib is None:
        path = os.path.join(_HERE, "_ref", "libref_hnsw.so")
        if not os.path.exists(path):
            import subprocess
            subprocess.run(["make", "-s", "hnsw_shim"], cwd=_HERE, check=True)
        if not os.path.exists(path):
            raise RuntimeError("oracle/_ref/libref_hnsw.so unavailable")
        L = ct.CDLL(path)
        L.ref_hnsw_create.restype = ct.c_void_p
        L.ref_hnsw_create.argtypes = [ct.c_size_t, ct.c_int, ct.c_size_t,
                                      ct.c_size_t, ct.c_size_t, ct.c_size_t,
                                      ct.c_size_t, ct.c_void_p]
        L.ref_hnsw_add_batch.restype = ct.c_int
        L.ref_hnsw_add_batch.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_void_p,
                                         ct.c_size_t, ct.c_size_t, ct.c_size_t,
                                         ct.c_void_p]
        L.ref_hnsw_search.restype = ct.c_size_t
        L.ref_hnsw_search.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_size_t,
                                      ct.c_size_t, ct.c_void_p, ct.c_void_p,
                                      ct.c_void_p]
        L.ref_hnsw_size.restype = ct.c_size_t
        L.ref_hnsw_size.argtypes = [ct.c_void_p]
        L.ref_hnsw_max_level.restype = ct.c_int
        L.ref_hnsw_max_level.argtypes = [ct.c_void_p]
        L.ref_hnsw_entry_slot.restype = ct.c_uint64
        L.ref_hnsw_entry_slot.argtypes = [ct.c_void_p]
        L.ref_hnsw_key_of_slot.restype = ct.c_uint64
        L.ref_hnsw_key_of_slot.argtypes = [ct.c_void_p, ct.c_uint32]
        L.ref_hnsw_node_level.restype = ct.c_int
        L.ref_hnsw_node_level.argtypes = [ct.c_void_p, ct.c_uint32]
        L.ref_hnsw_neighbors.restype = ct.c_size_t
        L.ref_hnsw_neighbors.argtypes = [ct.c_void_p, ct.c_uint32, ct.c_int,
                                         ct.c_void_p, ct.c_size_t]
        L.ref_hnsw_get_vector.restype = ct.c_int
        L.ref_hnsw_get_vector.argtypes = [ct.c_void_p, ct.c_uint32,
                                          ct.c_void_p, ct.c_size_t]
        L.ref_hnsw_save.restype = ct.c_int
        L.ref_hnsw_save.argtypes = [ct.c_void_p, ct.c_char_p, ct.c_void_p]
        L.ref_hnsw_load.restype = ct.c_void_p
        L.ref_hnsw_load.argtypes = [ct.c_char_p, ct.c_void_p]
        L.ref_hnsw_search_filtered.restype = ct.c_size_t
        L.ref_hnsw_search_filtered.argtypes = [ct.c_void_p, ct.c_void_p,
                                               ct.c_size_t, ct.c_size_t,
                                               ct.c_void_p, ct.c_uint64,
                                               ct.c_void_p, ct.c_void_p,
                                               ct.c_void_p]
        L.ref_hnsw_export_bulk.restype = ct.c_int64
        L.ref_hnsw_export_bulk.argtypes = [ct.c_void_p, ct.c_size_t,
                                           ct.c_void_p, ct.c_void_p,
                                           ct.c_void_p, ct.c_void_p,
                                           ct.c_void_p, ct.c_int64]
        L.ref_hnsw_export_level.restype = ct.c_int64
        L.ref_hnsw_export_level.argtypes = [ct.c_void_p, ct.c_int,
                                            ct.c_void_p, ct.c_void_p,
                                            ct.c_int64]
        L.ref_hnsw_free.argtypes = [ct.c_void_p]
        _lib = L
    return _lib


class RefHnsw:
    """The reference usearch engine (build + search + export)."""

    def __init__(self, dim, metric=METRIC_COS, connectivity=32,
                 expansion_add=128, expansion_search=64, capacity=0,
                 threads=1):
        err = ct.c_char_p(None)
        self.dim = dim
        self.metric = metric
        self._h = lib().ref_hnsw_create(dim, metric, connectivity,
                                        expansion_add, expansion_search,
                                        capacity, threads, ct.byref(err))
        if not self._h:
            raise RuntimeError(err.value)
        self.threads = threads

    def add(self, vecs, keys=None):
        vecs = np.ascontiguousarray(vecs, dtype=np.float32)
        n = vecs.shape[0]
        if keys is None:
            keys = np.arange(n, dtype=np.uint64)
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        err = ct.c_char_p(None)
        rc = lib().ref_hnsw_add_batch(self._h, keys.ctypes.data,
                                      vecs.ctypes.data, n, self.dim,
                                      self.threads, ct.byref(err))
        if rc != 0:
            raise RuntimeError(err.value)

    def search(self, queries, ef, k):
        queries = np.ascontiguousarray(queries, dtype=np.float32)
        nq = queries.shape[0]
        out_k = np.zeros((nq, k), dtype=np.uint64)
        out_d = np.full((nq, k), np.float32(np.finfo(np.float32).max))
        counts = np.zeros(nq, dtype=np.int64)
        err = ct.c_char_p(None)
        for i in range(nq):
            counts[i] = lib().ref_hnsw_search(
                self._h, queries[i].ctypes.data, ef, k,
                out_k[i].ctypes.data, out_d[i].ctypes.data, ct.byref(err))
            if err.value:
                raise RuntimeError(err.value)
        ids = out_k.astype(np.int64)
        for i in range(nq):
            ids[i, counts[i]:] = -1
        return ids, out_d

    def search_filtered(self, queries, ef, k, key_bits: np.ndarray):
        """usearch filtered_search with a key-bitset predicate (the
        usearchex membership-callback shape, cgo/usearchex.c:27-70)."""
        queries = np.ascontiguousarray(queries, dtype=np.float32)
        key_bits = np.ascontiguousarray(key_bits, dtype=np.uint32)
        nbits = key_bits.size * 32
        nq = queries.shape[0]
        out_k = np.zeros((nq, k), dtype=np.uint64)
        out_d = np.full((nq, k), np.float32(np.finfo(np.float32).max))
        err = ct.c_char_p(None)
        ids = np.full((nq, k), -1, dtype=np.int64)
        for i in range(nq):
            cnt = lib().ref_hnsw_search_filtered(
                self._h, queries[i].ctypes.data, ef, k, key_bits.ctypes.data,
                nbits, out_k[i].ctypes.data, out_d[i].ctypes.data,
                ct.byref(err))
            if err.value:
                raise RuntimeError(err.value)
            ids[i, :cnt] = out_k[i, :cnt].astype(np.int64)
        return ids, out_d

    def export_graph(self) -> "HnswGraph":
        L = lib()
        n = L.ref_hnsw_size(self._h)
        max_level = L.ref_hnsw_max_level(self._h)
        entry = L.ref_hnsw_entry_slot(self._h)
        levels = np.empty(n, dtype=np.int32)
        keys = np.empty(n, dtype=np.uint64)
        vecs = np.empty((n, self.dim), dtype=np.float32)
        # bulk C-side export (per-node ctypes calls do not scale past ~1e5)
        cap = np.int64(n) * 128 + 64
        l0_offs = np.empty(n + 1, dtype=np.int64)
        l0_nbrs = np.empty(cap, dtype=np.uint32)
        nnz = L.ref_hnsw_export_bulk(self._h, self.dim, levels.ctypes.data,
                                     keys.ctypes.data, vecs.ctypes.data,
                                     l0_offs.ctypes.data, l0_nbrs.ctypes.data,
                                     cap)
        if nnz < 0:
            raise RuntimeError("l0 adjacency overflow")
        adj = [(l0_offs, l0_nbrs[:nnz].copy())]
        for lvl in range(1, max_level + 1):
            offs = np.empty(n + 1, dtype=np.int64)
            nbrs = np.empty(cap, dtype=np.uint32)
            nz = L.ref_hnsw_export_level(self._h, lvl, offs.ctypes.data,
                                         nbrs.ctypes.data, cap)
            if nz < 0:
                raise RuntimeError("upper adjacency overflow")
            adj.append((offs, nbrs[:nz].copy()))
        return HnswGraph(self.dim, self.metric, int(entry), int(max_level),
                         levels, keys, vecs, adj)

    def save(self, path: str):
        err = ct.c_char_p(None)
        if lib().ref_hnsw_save(self._h, path.encode(), ct.byref(err)) != 0:
            raise RuntimeError(err.value)

    def __del__(self):
        try:
            if self._h:
                lib().ref_hnsw_free(self._h)
        except Exception:
            pass


@dataclass
class HnswGraph:
    dim: int
    metric: int
    entry_slot: int
    max_level: int
    levels: np.ndarray     # [n] int32
    keys: np.ndarray       # [n] uint64 (slot -> key)
    vecs: np.ndarray       # [n][dim] f32 (stored vectors)
    adj: list              # per level: (offsets[n+1] int64, neighbors u32)

    def neighbors(self, slot, level):
        offs, nbr = self.adj[level]
        return nbr[offs[slot]:offs[slot + 1]]


def _usearch_dist(metric, a, b):
    """f32 arithmetic like SimSIMD's serial f32 kernels (within 1e-5)."""
    a = a.astype(np.float32)
    b = b.astype(np.float32)
    if metric == METRIC_L2SQ:
        d = a - b
        return np.float32(np.dot(d, d))
    if metric == METRIC_IP:
        return np.float32(1.0) - np.float32(np.dot(a, b))
    ab = np.float32(np.dot(a, b))
    a2 = np.float32(np.dot(a, a))
    b2 = np.float32(np.dot(b, b))
    if a2 == 0 or b2 == 0:
        # usearch cos_normalize/guard: zero norm -> distance 1
        return np.float32(1.0)
    return np.float32(1.0 - ab / (np.sqrt(a2) * np.sqrt(b2)))


def oracle_search(g: HnswGraph, query: np.ndarray, ef: int, k: int,
                  allowed=None):
    """The numpy restatement (see module docstring for cites)."""
    q = query.astype(np.float32)

    def dist(slot):
        return float(_usearch_dist(g.metric, g.vecs[slot], q))

    # greedy descent: max_level .. 1 (search_for_one_, index.hpp:4466)
    closest = g.entry_slot
    closest_d = dist(closest)
    for level in range(g.max_level, 0, -1):
        changed = True
        while changed:
            changed = False
            for nb in g.neighbors(closest, level):
                d = dist(int(nb))
                if d < closest_d:
                    closest_d = d
                    closest = int(nb)
                    changed = True

    # base-level beam (search_to_find_in_base_, index.hpp:4688). `allowed`
    # (a key predicate) gates TOP insertion only — traversal pushes to next
    # under the same (top.size < ef or d < radius) condition either way
    # (index.hpp:4712-4752).
    def passes(slot):
        return allowed is None or allowed(int(g.keys[slot]))

    visited = {closest}
    nxt = [(closest_d, closest)]          # min-heap
    top = []                              # bounded max-heap (size <= ef)
    radius = closest_d
    if passes(closest):
        top = [(-closest_d, closest)]
    while nxt:
        cd, c = nxt[0]
        if cd > radius and len(top) == ef:
            break
        heapq.heappop(nxt)
        for nb in g.neighbors(c, 0):
            nb = int(nb)
            if nb in visited:
                continue
            visited.add(nb)
            d = dist(nb)
            if len(top) < ef or d < radius:
                heapq.heappush(nxt, (d, nb))
                if passes(nb):
                    if len(top) == ef:
                        heapq.heappushpop(top, (-d, nb))
                    else:
                        heapq.heappush(top, (-d, nb))
                    radius = -top[0][0]
    out = sorted((-d, s) for d, s in top)[:k]
    ids = np.full(k, -1, dtype=np.int64)
    dists = np.full(k, np.finfo(np.float32).max, dtype=np.float32)
    for i, (d, s) in enumerate(out):
        ids[i] = g.keys[s]
        dists[i] = d
    return ids, dists


def oracle_search_filtered_batch(g: HnswGraph, query: np.ndarray, ef: int,
                                 k: int, allowed):
    """The GPU filtered kernel's SPEC: usearch's predicate semantics with two
    bounded-memory deviations — the radius is recomputed once per EXPANSION
    (usearch tightens it incrementally inside the neighbor loop), and the
    candidate list is capped at 2*ef. Returned ids always satisfy the filter;
    id agreement vs usearch's filtered_search measured ~0.90-0.93 at ef=48
    and 30% selectivity with equivalent recall (see tests)."""
    q = query.astype(np.float32)

    def dist(slot):
        return float(_usearch_dist(g.metric, g.vecs[slot], q))

    cur = g.entry_slot
    cur_d = dist(cur)
    for level in range(g.max_level, 0, -1):
        changed = True
        while changed:
            changed = False
            best, bi = cur_d, -1
            for nb in g.neighbors(cur, level):
                d = dist(int(nb))
                if d < best:
                    best, bi = d, int(nb)
            if bi >= 0:
                cur, cur_d, changed = bi, best, True

    efc = 2 * ef
    lst = [(dist(cur), cur, bool(allowed(int(g.keys[cur]))), False)]
    visited = {cur}
    while True:
        pd = [d for d, s_, p, e in lst if p]
        pass_cnt = len(pd)
        radius = pd[ef - 1] if pass_cnt >= ef else float("inf")
        bi = next((i for i, t in enumerate(lst) if not t[3]), None)
        if bi is None:
            break
        if pass_cnt >= ef and lst[bi][0] > radius:
            break
        d0, s0, p0, _ = lst[bi]
        lst[bi] = (d0, s0, p0, True)
        new = []
        for nb in g.neighbors(s0, 0):
            nb = int(nb)
            if nb in visited:
                continue
            visited.add(nb)
            d = dist(nb)
            if pass_cnt >= ef and d >= radius:
                continue
            new.append((d, nb, bool(allowed(int(g.keys[nb]))), False))
        lst = sorted(lst + new, key=lambda t: t[0])[:efc]
    out = [(d, s_) for d, s_, p, e in lst if p][:k]
    ids = np.full(k, -1, dtype=np.int64)
    dists = np.full(k, np.finfo(np.float32).max, dtype=np.float32)
    for i, (d, s_) in enumerate(out):
        ids[i] = g.keys[s_]
        dists[i] = np.float32(d)
    return ids, dists
