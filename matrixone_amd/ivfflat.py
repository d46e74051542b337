"""IvfflatSearch — the MatrixOne-side host mirror of the IVF-Flat search
implementation, backed by the MI355X C-ABI instead of SQL + SimSIMD.

Mirrors pkg/vectorindex/cache/cache.go:65-84 VectorIndexSearchIf
{Search, SearchFloat32, Load, Destroy} and pkg/vectorindex/ivfflat/search.go
IvfflatSearch semantics: the index metric is derived from op_type
(types.go:141-147: vector_l2_ops -> Metric_L2sqDistance), the query's
original function name decides the sqrt transform (DistanceTransformIvfflat,
types.go:245-251), and RuntimeConfig carries Limit/Probe (types.go:268-323).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

import numpy as np

from . import engine

# op_type -> index metric (pkg/vectorindex/metric/types.go:141-147)
OP_TYPE_METRIC = {
    "vector_l2_ops": "l2sq",
    "vector_l2sq_ops": "l2sq",
    "vector_ip_ops": "ip",
    "vector_cosine_ops": "cos",
    "vector_l1_ops": "l1",
}

# SQL function names (metric/types.go DistFn_*)
DISTFN_L2 = "l2_distance"
DISTFN_L2SQ = "l2_distance_sq"
DISTFN_IP = "inner_product"
DISTFN_COS = "cosine_distance"
DISTFN_L1 = "l1_distance"


@dataclass
class IndexConfig:
    """pkg/vectorindex/types.go:102-160 IndexConfig.Ivfflat subset."""
    dimensions: int
    lists: int
    op_type: str = "vector_l2_ops"
    version: int = 0


@dataclass
class RuntimeConfig:
    """pkg/vectorindex/types.go:268-323 RuntimeConfig subset."""
    limit: int = 10
    probe: int = 1
    orig_func_name: str = DISTFN_L2
    nthreads: int = 0


@dataclass
class IndexSource:
    """Stand-in for the hidden-table reads (centroids + entries) that
    pkg/vectorindex/ivfflat/search.go:71-161 LoadCentroids and :917-1005
    scanEntries perform through the SQL executor. The embedding host hands
    the arrays straight from its storage scan."""
    centroids: np.ndarray           # [nlist][dim] f32
    vectors: np.ndarray             # [n][dim] f32 (entry vectors)
    assignments: np.ndarray         # [n] int32 centroid slot of each entry
    ids: Optional[np.ndarray] = None  # [n] int64 entry PKs


class IvfflatSearch:
    """VectorIndexSearchIf implementation (GPU-backed IvfflatSearch)."""

    def __init__(self, idxcfg: IndexConfig, source: IndexSource,
                 device: int = 0):
        self.idxcfg = idxcfg
        self.source = source
        self.device = device
        self.index: Optional[engine.IvfFlatIndex] = None

    # --- VectorIndexSearchIf ------------------------------------------------
    def Load(self, proc=None) -> None:
        """LoadIndex (ivfflat/search.go:163-198): centroids + entries into
        the device-resident index."""
        metric = OP_TYPE_METRIC[self.idxcfg.op_type]
        src = self.source
        n = src.vectors.shape[0]
        ix = engine.IvfFlatIndex(self.idxcfg.dimensions, self.idxcfg.lists,
                                 metric=metric, capacity=n,
                                 device=self.device)
        ix.add(src.vectors, ids=src.ids)
        ix.set_centroids(src.centroids)
        ix.set_assignments(src.assignments)
        ix.build()
        self.index = ix

    def _metric_for(self, rt: RuntimeConfig) -> str:
        """DistanceTransformIvfflat: sqrt only when the query's function is
        l2_distance over an l2sq index (metric/types.go:245-251)."""
        m = OP_TYPE_METRIC[self.idxcfg.op_type]
        if m == "l2sq" and rt.orig_func_name == DISTFN_L2:
            return "l2"
        return m

    def _require_loaded(self) -> engine.IvfFlatIndex:
        if self.index is None:
            raise RuntimeError("index not loaded — call Load() first")
        return self.index

    def Search(self, proc, query, rt: RuntimeConfig):
        """Returns (keys, distances) like ivfflat/search.go:702 Search —
        keys int64 array, distances float64, ascending."""
        ix = self._require_loaded()
        want = self._metric_for(rt)
        q = np.atleast_2d(np.asarray(query, dtype=np.float32))
        ids, dists = ix.search(q, rt.limit, rt.probe)
        d64 = dists.astype(np.float64)
        if want == "l2" and ix.metric == "l2sq":
            valid = ids >= 0
            d64 = np.where(valid, np.sqrt(d64, where=d64 >= 0, out=d64.copy()), d64)
        return ids, d64

    def SearchFloat32(self, proc, query, rt: RuntimeConfig,
                      out_keys: np.ndarray, out_dists: np.ndarray) -> None:
        """cache.go:70-72: write into caller-provided slices."""
        ids, d64 = self.Search(proc, query, rt)
        n = ids.size
        out_keys[:n] = ids.reshape(-1)
        out_dists[:n] = d64.reshape(-1).astype(np.float32)

    def Destroy(self) -> None:
        if self.index is not None:
            self.index.close()
            self.index = None
