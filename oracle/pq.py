"""ORACLE — test infrastructure ONLY.

IVF-PQ restatement (BASELINE.json config 5). The reference
serves IVF-PQ through cuVS (`cgo/cuvs/ivf_pq.hpp` wrapping
cuvs::neighbors::ivf_pq — an external dependency whose device code is NOT in
/root/reference, pinned by the reference at cuVS/RAFT 24.x via its cgo
build). Per the third-party rule, this module restates the PUBLISHED IVF-PQ
algorithm the reference's wrapper drives (Jégou, Douze, Schmid: "Product
Quantization for Nearest Neighbor Search", IEEE TPAMI 2011 — ADC over
per-subspace codebooks with coarse-residual encoding, the cuVS
`codebook_kind = PER_SUBSPACE` default), and parity for the GPU kernel is
anchored on THIS restatement plus the recall harness shape of the
reference's own pkg/cuvs/recall_test.go:38-75.

Pipeline:
  train:  residual r_i = x_i - centroid[assign_i]; per subspace m,
          k-means(256) over {r_i[m*sd:(m+1)*sd]}.
  encode: code_i[m] = argmin_j ||r_i_m - cb[m][j]||^2
  search: rank coarse centroids (f32, same as IVF-Flat); per (query, probed
          list): residual q' = q - centroid[l], LUT[m][j] = ||q'_m - cb[m][j]||^2,
          dist(i) = sum_m LUT[m][code_i[m]]; ascending top-k.
L2 only (config 5 is L2).
"""

from __future__ import annotations

import numpy as np


def train_pq(vecs: np.ndarray, centroids: np.ndarray, assign: np.ndarray,
             nsub: int, iters: int = 10, seed: int = 0):
    """Per-subspace 256-entry codebooks over coarse residuals.
    Returns cb [nsub][256][sd] f32."""
    n, d = vecs.shape
    assert d % nsub == 0
    sd = d // nsub
    res = vecs - centroids[assign]
    rng = np.random.Generator(np.random.PCG64(seed))
    cb = np.empty((nsub, 256, sd), dtype=np.float32)
    for m in range(nsub):
        sub = res[:, m * sd:(m + 1) * sd].astype(np.float32)
        idx = rng.choice(n, size=min(n, 256), replace=False)
        c = sub[idx].copy()
        if len(c) < 256:
            c = np.concatenate([c, rng.standard_normal(
                (256 - len(c), sd)).astype(np.float32) * 1e-3 + c.mean(0)])
        for _ in range(iters):
            d2 = ((sub[:, None, :] - c[None]) ** 2).sum(-1) if n <= 20000 \
                else None
            if d2 is None:
                # chunked for big n
                a = np.empty(n, dtype=np.int64)
                for c0 in range(0, n, 16384):
                    blk = sub[c0:c0 + 16384]
                    a[c0:c0 + 16384] = ((blk[:, None, :] - c[None]) ** 2
                                        ).sum(-1).argmin(1)
            else:
                a = d2.argmin(1)
            for j in range(256):
                mask = a == j
                if mask.any():
                    c[j] = sub[mask].mean(0)
        cb[m] = c
    return cb


def encode_pq(vecs: np.ndarray, centroids: np.ndarray, assign: np.ndarray,
              cb: np.ndarray) -> np.ndarray:
    """codes [n][nsub] uint8."""
    n, d = vecs.shape
    nsub, _, sd = cb.shape
    res = (vecs - centroids[assign]).astype(np.float32)
    codes = np.empty((n, nsub), dtype=np.uint8)
    for m in range(nsub):
        sub = res[:, m * sd:(m + 1) * sd]
        for c0 in range(0, n, 16384):
            blk = sub[c0:c0 + 16384]
            d2 = ((blk[:, None, :] - cb[m][None]) ** 2).sum(-1)
            codes[c0:c0 + 16384, m] = d2.argmin(1).astype(np.uint8)
    return codes


class PqIndex:
    """CSR IVF-PQ index (same list structure as oracle.IvfIndex)."""

    def __init__(self, centroids, cb, codes, assign, ids=None):
        self.centroids = np.ascontiguousarray(centroids, dtype=np.float32)
        self.cb = np.ascontiguousarray(cb, dtype=np.float32)
        self.codes = np.ascontiguousarray(codes, dtype=np.uint8)
        self.ids = None if ids is None else np.asarray(ids, np.int64)
        nlist = self.centroids.shape[0]
        assign = np.asarray(assign, dtype=np.int64)
        order = np.argsort(assign, kind="stable")
        counts = np.bincount(assign, minlength=nlist)
        self.list_offsets = np.zeros(nlist + 1, dtype=np.int64)
        np.cumsum(counts, out=self.list_offsets[1:])
        self.slot_rows = order.astype(np.int64)

    def search(self, queries: np.ndarray, probe: int, k: int):
        """ADC search; f32 LUT accumulation in subspace order (the order the
        GPU kernel also uses). Returns ids/dists like IvfIndex.search."""
        queries = np.ascontiguousarray(queries, dtype=np.float32)
        nq, d = queries.shape
        nsub, _, sd = self.cb.shape
        nlist = self.centroids.shape[0]
        probe = min(probe, nlist)
        out_ids = np.full((nq, k), -1, dtype=np.int64)
        out_d = np.full((nq, k), np.float32(np.finfo(np.float32).max))
        for qi in range(nq):
            q = queries[qi]
            cd = ((self.centroids.astype(np.float64) -
                   q.astype(np.float64)) ** 2).sum(1)
            ranked = np.argsort(cd, kind="stable")[:probe]
            cand_d = []
            cand_id = []
            for l in ranked:
                lo, hi = self.list_offsets[l], self.list_offsets[l + 1]
                if hi == lo:
                    continue
                rows = self.slot_rows[lo:hi]
                resid = (q - self.centroids[l]).astype(np.float32)
                lut = np.empty((nsub, 256), dtype=np.float32)
                for m in range(nsub):
                    diffs = resid[m * sd:(m + 1) * sd][None, :] - self.cb[m]
                    lut[m] = (diffs.astype(np.float32) ** 2).sum(1)
                codes = self.codes[rows]  # [cnt][nsub]
                # accumulate in subspace order, f32 (matches the kernel)
                dist = np.zeros(len(rows), dtype=np.float32)
                for m in range(nsub):
                    dist += lut[m][codes[:, m]]
                cand_d.append(dist)
                cand_id.append(rows)
            if not cand_d:
                continue
            dists = np.concatenate(cand_d)
            rows = np.concatenate(cand_id)
            order = np.lexsort((rows, dists))[:k]
            sel = rows[order]
            out_ids[qi, :len(order)] = (self.ids[sel] if self.ids is not None
                                        else sel)
            out_d[qi, :len(order)] = dists[order]
        return out_ids, out_d
