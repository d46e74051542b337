"""Sharded multi-GPU search: one process per GPU over torch.distributed
(RCCL on ROCm), replacing the reference's in-process device pool + CPU merge
(cgo/cuvs/index_base.hpp SHARDED mode + helper.h:137-167
cpu_topk_merge_sharded) with an RCCL all-gather of per-shard top-k blocks
over xGMI and a GPU-side merge.

The exchange per batch is one all_gather of [B, k] (id i64, dist f32) pairs
(~B*k*12 B per shard — far below the xGMI per-link knee, latency-dominated,
so a single-shot all-gather, not a ring reduction).

Shard layout: each rank owns a disjoint row range / its own IVF lists
(embarrassingly parallel shard search, as the reference's SHARDED mode).
REPLICATED mode (types.go:73-98) = every rank holds the full index and serves
its own query stream — no exchange; bench uses it for throughput scaling.
"""

from __future__ import annotations

import numpy as np

try:
    import torch
    import torch.distributed as dist
except Exception:  # pragma: no cover - torch is present in all target envs
    torch = None
    dist = None


def merge_topk_torch(shard_ids, shard_dists, k: int):
    """Merge per-shard dense top-k blocks — semantics of
    cpu_topk_merge_sharded (helper.h:137-167): skip -1 sentinels, dense
    ascending top-k, pad (-1, FLT_MAX). Runs on whatever device the tensors
    live on (GPU after an RCCL all-gather).

    shard_ids:  [nshard, nq, k] int64
    shard_dists:[nshard, nq, k] float32
    """
    nshard, nq, kk = shard_ids.shape
    ids = shard_ids.permute(1, 0, 2).reshape(nq, nshard * kk)
    dists = shard_dists.permute(1, 0, 2).reshape(nq, nshard * kk).clone()
    fmax = torch.finfo(torch.float32).max
    dists[ids < 0] = fmax
    order = dists.argsort(dim=1, stable=True)[:, :k]
    out_d = torch.gather(dists, 1, order)
    out_i = torch.gather(ids, 1, order)
    out_i[out_d >= fmax] = -1
    return out_i, out_d


def allgather_merge(local_ids: np.ndarray, local_dists: np.ndarray, k: int,
                    device=None, group=None):
    """All-gather each rank's [nq, k] block and merge. Works on the 'nccl'
    (=RCCL) backend with CUDA tensors and on 'gloo' with CPU tensors (the
    world_size-2 CPU test path)."""
    assert dist is not None and dist.is_initialized()
    world = dist.get_world_size(group)
    backend = dist.get_backend(group)
    use_cuda = backend == "nccl"
    dev = device if device is not None else (
        torch.device("cuda") if use_cuda else torch.device("cpu"))
    t_ids = torch.from_numpy(np.ascontiguousarray(local_ids)).to(dev)
    t_dists = torch.from_numpy(np.ascontiguousarray(local_dists)).to(dev)
    gather_ids = [torch.empty_like(t_ids) for _ in range(world)]
    gather_dists = [torch.empty_like(t_dists) for _ in range(world)]
    dist.all_gather(gather_ids, t_ids, group=group)
    dist.all_gather(gather_dists, t_dists, group=group)
    all_ids = torch.stack(gather_ids)
    all_dists = torch.stack(gather_dists)
    out_i, out_d = merge_topk_torch(all_ids, all_dists, k)
    return out_i.cpu().numpy(), out_d.cpu().numpy()
