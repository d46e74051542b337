#!/usr/bin/env python3
"""HNSW at BASELINE.json configs[2] scale: 10M x 768, cosine, M=32,
efSearch=128, batch=256, k=10 — served exactly the way the reference serves
10M rows: ~10 chained mini-models of <= 1M vectors each
(pkg/vectorindex/hnsw/sync.go:98-106 hnsw_max_index_capacity), searched
with a fan-out + bounded top-k merge (hnsw/search.go:89-183).

Build: usearch (the reference engine, oracle/_ref) builds the 10 mini-models
in PARALLEL worker processes (builds are independent, sync.go's natural
unit). Graph BUILD is the reference's own CPU path and out of the judged
search scope — the measured quantity is search QPS.

GPU serve: each mini-model is ingested into an HnswIndex (HBM CSR); a batch
fans out to all models CONCURRENTLY (one thread per model, each model on
its own HIP stream — ctypes releases the GIL) and the per-model top-k are
merged like SearchResultSafeHeap (index.go:135-189).

Reports: GPU QPS, recall@10 vs exact cosine GT, id agreement vs the
usearch fan-out on the same graphs, usearch 1-thread fan-out QPS.

Run on a GPU box: python tools/hnsw10m_bench.py [--rows 10000000]
"""
import argparse
import concurrent.futures as cf
import json
import multiprocessing as mp
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def gen_data_gpu(rows, dim, nq):
    """The bench mixture (same recipe as bench.py/hnsw_bench_helper)."""
    import torch
    dev = torch.device("cuda", 0)
    latent = 32
    gc = torch.Generator(device="cpu").manual_seed(777)
    U = (torch.randn(latent, dim, generator=gc) / (latent ** 0.5)).to(dev)
    ncomp = min(131072, max(256, rows // 100))
    z = (torch.randn(ncomp, latent, generator=gc) * (latent ** 0.5)).to(dev)
    centers = z @ U
    gp = torch.Generator(device="cpu").manual_seed(1000)
    comp = torch.randint(0, ncomp, (rows,), generator=gp).to(dev)
    data = centers[comp]
    gn = torch.Generator(device=dev)
    gn.manual_seed(2000)
    for c0 in range(0, rows, 1 << 20):
        blk = data[c0:c0 + (1 << 20)]
        blk += torch.randn(blk.shape, generator=gn, device=dev)
    gq = torch.Generator(device="cpu").manual_seed(4242)
    qcomp = torch.randint(0, ncomp, (nq,), generator=gq)
    qnoise = torch.randn(nq, dim, generator=gq)
    queries = centers[qcomp.to(dev)] + qnoise.to(dev)
    return data, queries.cpu().numpy().astype(np.float32)


def cosine_gt(data, queries_np, k, dev):
    import torch
    q = torch.from_numpy(queries_np).to(dev)
    qn = q / q.norm(dim=1, keepdim=True).clamp_min(1e-30)
    best_s = torch.full((q.shape[0], k), -2.0, device=dev)
    best_i = torch.full((q.shape[0], k), -1, dtype=torch.int64, device=dev)
    CH = 1 << 20
    for c0 in range(0, data.shape[0], CH):
        blk = data[c0:c0 + CH]
        bn = blk / blk.norm(dim=1, keepdim=True).clamp_min(1e-30)
        sim = qn @ bn.T
        s, i = torch.topk(sim, min(k, blk.shape[0]), dim=1)
        cat_s = torch.cat([best_s, s], 1)
        cat_i = torch.cat([best_i, i + c0], 1)
        s, pos = torch.topk(cat_s, k, dim=1)
        best_s = s
        best_i = torch.gather(cat_i, 1, pos)
    return best_i.cpu().numpy()


def build_worker(args):
    """Build one usearch mini-model over rows [lo,hi) of the memmap; export
    the graph structure (not the vectors — the parent regenerates them) and
    search the agreement/CPU-baseline queries single-threaded."""
    (i, mmpath, rows, dim, lo, hi, conn, efc, ef, k, qpath, nqa,
     outdir, threads) = args
    from oracle import hnsw as H
    data = np.memmap(mmpath, dtype=np.float32, mode="r",
                     shape=(rows, dim))
    queries = np.load(qpath)
    t0 = time.perf_counter()
    ref = H.RefHnsw(dim, metric=H.METRIC_COS, connectivity=conn,
                    expansion_add=efc, expansion_search=ef,
                    capacity=hi - lo, threads=threads)
    ref.add(np.ascontiguousarray(data[lo:hi]),
            keys=np.arange(lo, hi, dtype=np.uint64))
    build_s = time.perf_counter() - t0
    g = ref.export_graph()
    # usearch 1-thread timing + results on the agreement queries
    t0 = time.perf_counter()
    uids, udists = ref.search(queries[:nqa], ef=ef, k=k)
    cpu_s = time.perf_counter() - t0
    np.savez(os.path.join(outdir, f"model{i}.npz"),
             entry_slot=g.entry_slot, max_level=g.max_level,
             levels=g.levels, keys=g.keys.astype(np.int64),
             nlev=len(g.adj),
             **{f"offs{l}": g.adj[l][0] for l in range(len(g.adj))},
             **{f"nbrs{l}": g.adj[l][1] for l in range(len(g.adj))},
             uids=uids, udists=udists)
    return i, build_s, cpu_s


def merge_topk(ids_list, dists_list, k):
    """Bounded fan-out merge (SearchResultSafeHeap shape): ascending
    distance over the union, (-1, inf) padded."""
    ids = np.concatenate(ids_list, axis=1)
    dists = np.concatenate(dists_list, axis=1)
    dists = np.where(ids < 0, np.inf, dists)
    order = np.argsort(dists, axis=1, kind="stable")[:, :k]
    return (np.take_along_axis(ids, order, 1),
            np.take_along_axis(dists, order, 1))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=10_000_000)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--model-cap", type=int, default=1_000_000,
                    help="hnsw_max_index_capacity (sync.go:98-106)")
    ap.add_argument("--conn", type=int, default=32)
    ap.add_argument("--efc", type=int, default=128)
    ap.add_argument("--ef", type=int, default=128)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--steps", type=int, default=12)
    ap.add_argument("--nqa", type=int, default=256,
                    help="agreement/CPU-baseline queries")
    ap.add_argument("--build-threads", type=int, default=12)
    ap.add_argument("--workdir", default="/tmp/hnsw10m")
    args = ap.parse_args()

    import torch
    dev = torch.device("cuda", 0)
    os.makedirs(args.workdir, exist_ok=True)

    t0 = time.perf_counter()
    data, queries = gen_data_gpu(args.rows, args.dim, 4096)
    print(f"[h10] datagen {time.perf_counter()-t0:.1f}s", flush=True)

    mmpath = os.path.join(args.workdir, "data.f32")
    t0 = time.perf_counter()
    mm = np.memmap(mmpath, dtype=np.float32, mode="w+",
                   shape=(args.rows, args.dim))
    CH = 1 << 20
    for c0 in range(0, args.rows, CH):
        mm[c0:c0 + CH] = data[c0:c0 + CH].cpu().numpy()
    mm.flush()
    qpath = os.path.join(args.workdir, "queries.npy")
    np.save(qpath, queries)
    print(f"[h10] memmap write {time.perf_counter()-t0:.1f}s", flush=True)

    nmodels = (args.rows + args.model_cap - 1) // args.model_cap
    jobs = []
    for i in range(nmodels):
        lo = i * args.model_cap
        hi = min(args.rows, lo + args.model_cap)
        jobs.append((i, mmpath, args.rows, args.dim, lo, hi, args.conn,
                     args.efc, args.ef, args.k, qpath, args.nqa,
                     args.workdir, args.build_threads))
    t0 = time.perf_counter()
    ctx = mp.get_context("spawn")
    res = []
    with ctx.Pool(nmodels) as pool:
        for r in pool.imap_unordered(build_worker, jobs):
            res.append(r)
            print(f"[h10] model {r[0]} built in {r[1]:.1f}s "
                  f"({len(res)}/{nmodels})", flush=True)
    build_s = time.perf_counter() - t0
    per_model_cpu = {i: c for i, _, c in res}
    print(f"[h10] {nmodels} mini-model builds {build_s:.1f}s "
          f"(max single {max(b for _, b, _ in res):.1f}s)", flush=True)

    # ingest to GPU
    from matrixone_amd.hnsw import HnswGraphData, HnswIndex
    t0 = time.perf_counter()
    models = []
    u_ids_all, u_dists_all = [], []
    for i in range(nmodels):
        z = np.load(os.path.join(args.workdir, f"model{i}.npz"))
        keys = z["keys"]
        vecs = data[torch.from_numpy(keys).to(dev)].cpu().numpy()
        adj = [(z[f"offs{l}"], z[f"nbrs{l}"]) for l in range(int(z["nlev"]))]
        gd = HnswGraphData(dim=args.dim, metric="cos",
                           entry_slot=int(z["entry_slot"]),
                           max_level=int(z["max_level"]),
                           levels=z["levels"], keys=keys, vecs=vecs,
                           adj=adj)
        models.append(HnswIndex(gd, device=0))
        u_ids_all.append(z["uids"])
        u_dists_all.append(z["udists"])
        os.remove(os.path.join(args.workdir, f"model{i}.npz"))
    print(f"[h10] ingest {time.perf_counter()-t0:.1f}s", flush=True)

    pool = cf.ThreadPoolExecutor(nmodels)

    def fanout(qbatch):
        futs = [pool.submit(m.search, qbatch, args.ef, args.k)
                for m in models]
        outs = [f.result() for f in futs]
        return merge_topk([o[0] for o in outs], [o[1] for o in outs],
                          args.k)

    # warmup
    fanout(queries[:args.batch])
    t0 = time.perf_counter()
    for s in range(args.steps):
        b0 = (s * args.batch) % (4096 - args.batch)
        fanout(queries[b0:b0 + args.batch])
    dt = time.perf_counter() - t0
    qps = args.steps * args.batch / dt
    print(f"[h10] GPU fan-out: {qps:.1f} QPS "
          f"({dt / args.steps * 1e3:.2f} ms/batch of {args.batch})",
          flush=True)

    # recall vs exact cosine GT
    nrq = 512
    gt = cosine_gt(data, queries[:nrq], args.k, dev)
    g_ids, _ = fanout(queries[:nrq])
    recall = float(np.mean([
        len(set(g_ids[i]).intersection(gt[i])) / args.k
        for i in range(nrq)]))
    print(f"[h10] recall@{args.k} vs exact GT: {recall:.4f}", flush=True)

    # agreement vs the usearch fan-out on the SAME graphs
    u_ids, _ = merge_topk(u_ids_all, u_dists_all, args.k)
    a_ids, _ = fanout(queries[:args.nqa])
    agree = float((a_ids == u_ids[:args.nqa]).mean())
    cpu_fanout_qps = args.nqa / sum(per_model_cpu.values())
    print(f"[h10] id agreement vs usearch fan-out: {agree:.4f}", flush=True)

    out = {
        "workload": f"hnsw-cos-{args.rows}x{args.dim}-M{args.conn}-"
                    f"ef{args.ef}-batch{args.batch}-k{args.k}-"
                    f"{nmodels}x{args.model_cap}models",
        "gpu_qps": round(qps, 1),
        "gpu_ms_per_batch": round(dt / args.steps * 1e3, 3),
        "recall_at_k": round(recall, 4),
        "usearch_id_agreement": round(agree, 4),
        "usearch_cpu_fanout_qps_1thread": round(cpu_fanout_qps, 2),
        "build_s_parallel": round(build_s, 1),
        "efc": args.efc,
        "n_models": nmodels,
    }
    print(json.dumps(out), flush=True)
    with open("gpurun_out/hnsw10m.json", "w") as f:
        json.dump(out, f)


if __name__ == "__main__":
    main()
