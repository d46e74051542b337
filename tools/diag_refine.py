#!/usr/bin/env python3
"""Diagnose the two-stage refine recall plateau at scale (round-2).

Observed: recall@10 = 0.9414 at R=64/128/256 identically on the 10M flagship
workload — R never affects the result, while a numpy replication of the byte
quantization says byte-rank<64 recall should be 1.0000 at 2M. This script
isolates which stage breaks:

  stage-1 (byte scan + top-R select): searched at k=R, the returned id set
    IS the stage-1 selection (exact re-rank preserves the set). Compare
    against a torch replication of the byte image's top-R.
  stage-2 (re-rank + final select): if top-10-of-k=R results reach ~1.0
    recall while k=10 search stays at 0.94, the stage-2/compose path is at
    fault.

Usage (GPU box): python tools/diag_refine.py --rows 2000000 [--refine 64]
"""
import argparse
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import bench  # reuse make_mixture / kmeans / assign / exact_gt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=2_000_000)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--nlist", type=int, default=4096)
    ap.add_argument("--nprobe", type=int, default=32)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--refine", type=int, default=128)
    ap.add_argument("--nrq", type=int, default=256)
    args = ap.parse_args()

    dev = torch.device("cuda", 0)
    torch.cuda.set_device(dev)
    from matrixone_amd import engine

    t0 = time.perf_counter()
    data, queries_host = bench.make_mixture(args.rows, args.dim, 1, 0, dev)
    cent = bench.kmeans_torch(data, args.nlist, iters=8, seed=123)
    assign = bench.assign_torch(data, cent)
    queries = queries_host.to(dev)
    torch.cuda.synchronize()
    print(f"[diag] data+kmeans {time.perf_counter()-t0:.1f}s", flush=True)

    ix = engine.IvfFlatIndex(args.dim, args.nlist, metric="l2sq",
                             capacity=args.rows, device=0)
    CH = 1 << 21
    for c0 in range(0, args.rows, CH):
        ix.add(data[c0:c0 + CH].cpu().numpy())
    ix.set_centroids(cent.cpu().numpy())
    ix.set_assignments(assign.cpu().numpy())
    ix.build()
    print(f"[diag] built {time.perf_counter()-t0:.1f}s", flush=True)

    nrq, k, R = args.nrq, args.k, args.refine
    gt = bench.exact_gt(data, queries[:nrq], k).cpu().numpy()
    q = queries[:nrq].contiguous()

    def recall(ids, gt_ids, kk):
        return float(np.mean([
            len(set(ids[i][:kk]).intersection(gt_ids[i][:kk])) / kk
            for i in range(ids.shape[0])]))

    # exact path
    e_ids, e_d = ix.search_device(q, k, args.nprobe)
    print(f"[diag] exact k={k}: recall {recall(e_ids, gt, k):.4f}")

    ix.enable_refine(R)
    r_ids, r_d = ix.search_device(q, k, args.nprobe)
    print(f"[diag] refine R={R} k={k}: recall {recall(r_ids, gt, k):.4f}")

    # stage-1 set: search with limit=R (stage-2 select becomes identity-ish)
    s_ids, s_d = ix.search_device(q, R, args.nprobe)
    print(f"[diag] refine R={R} k={R}: top-{k} recall "
          f"{recall(s_ids, gt, k):.4f}")
    cover = float(np.mean([
        len(set(s_ids[i]).intersection(gt[i])) / k for i in range(nrq)]))
    print(f"[diag] stage-1 set (k=R ids) covers GT: {cover:.4f}")

    # does k=10 result equal the exact re-rank of the k=R result's best 10?
    agree = float(np.mean([
        len(set(r_ids[i]).intersection(s_ids[i][:k])) / k
        for i in range(nrq)]))
    print(f"[diag] k={k} ids == first {k} of k=R ids: {agree:.4f}")

    # torch replication of the byte image (enable_refine's quantile recipe)
    rows_f32 = data  # slot order == add order
    ns = min(args.rows, 131072)
    samp = rows_f32[:ns].reshape(-1)
    nlo = samp.numel() // 1000
    vals, _ = torch.sort(samp)
    lo = float(vals[nlo])
    hi = float(vals[samp.numel() - 1 - nlo])
    mul = 255.0 / (hi - lo)
    add = -lo * mul - 128.0
    print(f"[diag] torch rq: lo={lo:.4f} hi={hi:.4f} mul={mul:.5f}")

    def quant(x):
        v = torch.round((x * mul).double() + add)
        return v.clamp(-128, 127).to(torch.float64)

    qq = quant(q)
    # byte distances of GT rows vs their byte-rank among ALL rows (f64 GEMM
    # in chunks; only a subset of queries to bound time)
    nsub = min(64, nrq)
    ranks = np.zeros((nsub, k), dtype=np.int64)
    gt_t = torch.from_numpy(gt[:nsub]).to(dev)
    gt_rows = quant(data[gt_t.reshape(-1)]).reshape(nsub, k, args.dim)
    d_gt = ((gt_rows - qq[:nsub, None, :]) ** 2).sum(-1)  # [nsub,k] f64
    below = torch.zeros((nsub, k), dtype=torch.int64, device=dev)
    CH2 = 1 << 19
    for c0 in range(0, args.rows, CH2):
        blk = quant(data[c0:c0 + CH2])
        d2 = (blk * blk).sum(1)[None, :] - 2.0 * (qq[:nsub] @ blk.T) \
            + (qq[:nsub] * qq[:nsub]).sum(1)[:, None]
        below += (d2[:, None, :] < d_gt[:, :, None]).sum(-1)
    ranks = below.cpu().numpy()
    for RR in (10, 64, 128, 256, 1024):
        rec = float((ranks < RR).mean())
        print(f"[diag] torch byte-rank<{RR}: {rec:.4f}")
    ix.close()


if __name__ == "__main__":
    main()
