#!/usr/bin/env python3
"""Benchmark: MatrixOne ANN vector-index search hot path on MI355X.

Workload (BASELINE.json configs[1], the metric's single-GPU operating point):
  IVF-Flat L2, 10M x 768 f32, nlist=4096, nprobe=32, batch=1024, k=10,
  recall@10 >= 0.95 verified against exact ground truth.

One "step" = one batch of 1024 queries through the full hot path
(centroid rank -> list scan -> top-k -> id gather), queries HBM-resident
when the timed region starts (device-pointer search).

Scaling (--gpus N via torch.distributed.run): WEAK — each rank owns its own
10M-row shard (lists sharded, SHARDED mode per cgo/cuvs/index_base.hpp:104),
every rank serves the same query batch, one RCCL all-gather of [B,k]
(id,dist) blocks over xGMI + GPU merge per step (replaces the reference's
CPU cpu_topk_merge_sharded, helper.h:137-167). value = queries/s of the
whole job over the N x 10M corpus.

Emits ONE JSON line from rank 0 including `roofline` (HIP-event time +
algorithmic bytes of the list-scan kernel vs 8 TB/s HBM peak) and
`cpu_baseline` (the oracle CPU port of the reference probe, OpenMP, timed on
a bounded sample of the same workload).
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

HBM_PEAK_GBS = 8000.0  # MI355X_MICROARCH.md: HBM3E 8.0 TB/s spec peak


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def make_mixture(rows, dim, world, rank, device):
    """Synthetic embedding-like corpus: a seeded LOW-RANK Gaussian mixture.
    Pure i.i.d. Gaussian vectors have no neighbor structure at d=768, and an
    ISOTROPIC full-rank mixture makes all non-own components equidistant so
    the ~32 smallest-norm "central" lists get probed by ~90% of queries (an
    artifact no real embedding set shows). Real embeddings live near a
    low-dimensional manifold: component centers are drawn in a latent
    32-dim subspace (shared across ranks), points add full-dim sigma=1.0
    noise. Queries are held-out draws from the same mixture, identical on
    all ranks."""
    latent, sigma = 32, 1.0
    ncomp = min(131072, max(256, rows * world // 100))
    gc = torch.Generator(device="cpu").manual_seed(777)
    U = (torch.randn(latent, dim, generator=gc) / (latent ** 0.5)).to(device)
    z = (torch.randn(ncomp, latent, generator=gc) * (latent ** 0.5)).to(device)
    centers = z @ U
    del z
    gp = torch.Generator(device="cpu").manual_seed(1000 + rank)
    comp = torch.randint(0, ncomp, (rows,), generator=gp).to(device)
    data = centers[comp]
    del comp
    gn = torch.Generator(device=device)
    gn.manual_seed(2000 + rank)
    for c0 in range(0, rows, 1 << 20):
        blk = data[c0:c0 + (1 << 20)]
        blk += sigma * torch.randn(blk.shape, generator=gn, device=device)
    gq = torch.Generator(device="cpu").manual_seed(4242)
    qcomp = torch.randint(0, ncomp, (10000,), generator=gq)
    qnoise = sigma * torch.randn(10000, dim, generator=gq)
    queries_host = centers[qcomp.to(device)].cpu() + qnoise
    del centers
    return data, queries_host


def kmeans_torch(data_gpu: torch.Tensor, nlist: int, iters: int, seed: int):
    """Harness-side k-means (index BUILD — out of the judged search path,
    SURVEY.md §2 kmeans row). f32 matmul Lloyd on the GPU."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    n = data_gpu.shape[0]
    sample = data_gpu[torch.randperm(n, generator=g)[:min(n, 262144)].to(data_gpu.device)]
    idx = torch.randperm(sample.shape[0], generator=g)[:nlist].to(data_gpu.device)
    cent = sample[idx].clone()
    for _ in range(iters):
        # assignment by expanded distance (build-time only)
        d2 = (sample * sample).sum(1, keepdim=True) \
            - 2.0 * (sample @ cent.T) + (cent * cent).sum(1)[None, :]
        a = d2.argmin(1)
        cent_new = torch.zeros_like(cent)
        counts = torch.zeros(nlist, device=cent.device, dtype=torch.float32)
        cent_new.index_add_(0, a, sample)
        counts.index_add_(0, a, torch.ones_like(a, dtype=torch.float32))
        empty = counts == 0
        counts = counts.clamp(min=1.0)
        cent = cent_new / counts[:, None]
        if empty.any():
            # reseed empty lists from random sample rows
            ridx = torch.randint(0, sample.shape[0], (int(empty.sum()),),
                                 device=cent.device)
            cent[empty] = sample[ridx]
    return cent.contiguous()


def assign_torch(data_gpu: torch.Tensor, cent: torch.Tensor,
                 chunk: int = 131072) -> torch.Tensor:
    """Nearest-centroid assignment, chunked f32 matmul (build-time)."""
    n = data_gpu.shape[0]
    out = torch.empty(n, dtype=torch.int32, device=data_gpu.device)
    cnorm = (cent * cent).sum(1)
    for c0 in range(0, n, chunk):
        blk = data_gpu[c0:c0 + chunk]
        d2 = (blk * blk).sum(1, keepdim=True) - 2.0 * (blk @ cent.T) + cnorm[None, :]
        out[c0:c0 + chunk] = d2.argmin(1).to(torch.int32)
    return out


def exact_gt(data_gpu: torch.Tensor, queries: torch.Tensor, k: int,
             chunk: int = 1 << 20):
    """Exact L2 top-k ground truth (recall measurement only)."""
    nq = queries.shape[0]
    qnorm = (queries * queries).sum(1, keepdim=True)
    best_d = torch.full((nq, k), float("inf"), device=data_gpu.device)
    best_i = torch.full((nq, k), -1, dtype=torch.int64, device=data_gpu.device)
    for c0 in range(0, data_gpu.shape[0], chunk):
        blk = data_gpu[c0:c0 + chunk]
        d2 = qnorm - 2.0 * (queries @ blk.T) + (blk * blk).sum(1)[None, :]
        d, i = torch.topk(d2, min(k, blk.shape[0]), dim=1, largest=False)
        cat_d = torch.cat([best_d, d], 1)
        cat_i = torch.cat([best_i, i + c0], 1)
        d, pos = torch.topk(cat_d, k, dim=1, largest=False)
        best_d = d
        best_i = torch.gather(cat_i, 1, pos)
    return best_i


def run_cpu_baseline(host_vecs: np.ndarray, cent: np.ndarray,
                     assign: np.ndarray, queries: np.ndarray, nprobe: int,
                     k: int, sample_q: int):
    """The oracle (CPU port of the reference probe: rankCentroids + list scan
    + top-k, OpenMP over queries) on a bounded sample of the same workload."""
    from oracle import oracle as orc
    idx = orc.IvfIndex(cent, host_vecs, assign.astype(np.int64))
    idx.search(orc.METRIC_L2SQ, queries[:2], nprobe, k)  # warm
    # grow the sample until >= ~8 s of CPU wall (bounded at 1024 queries)
    nq, dt = sample_q, 0.0
    while True:
        t0 = time.perf_counter()
        idx.search(orc.METRIC_L2SQ, queries[:nq], nprobe, k)
        dt = time.perf_counter() - t0
        if dt >= 8.0 or nq >= min(1024, queries.shape[0]):
            break
        nq = min(min(1024, queries.shape[0]),
                 max(nq * 2, int(nq * 10.0 / max(dt, 0.1))))
    cores = orc.lib().moann_oracle_num_threads()
    return {
        "value": round(nq / dt, 3),
        "unit": "queries/s",
        "cores": int(cores),
        "kind": "port",
        "sample": f"{nq} queries of the same per-shard probe "
                  f"(nprobe={nprobe}, k={k}), {dt:.2f}s wall",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows", type=int, default=10_000_000,
                    help="rows PER GPU (weak scaling)")
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--nlist", type=int, default=4096)
    ap.add_argument("--nprobe", type=int, default=32)
    ap.add_argument("--batch", type=int, default=1024)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--recall-queries", type=int, default=256)
    ap.add_argument("--cpu-baseline-queries", type=int, default=48)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--refine", type=int, default=64,
                    help="two-stage scan: byte first pass + exact f32 "
                         "re-rank of the top-R (0 = exact one-pass scan). "
                         "GPU-validated at 10M post the QT-8 job-geometry "
                         "fix: R=64/128/256 all recall 1.0000 "
                         "(profiles/r02_refine_sweep.log; R=64 best at steps 20). "
                         "ADAPTIVE: if the measured recall@k gate (0.95) "
                         "fails with refine on, the bench falls back to "
                         "the exact one-pass scan and re-measures — the "
                         "reported configuration always satisfies the "
                         "gate if the exact path does")
    ap.add_argument("--qtype", default="f32",
                    choices=["f32", "f16", "bf16", "int8", "uint8"],
                    help="storage type; the BASELINE metric is f32 — narrow "
                         "runs are secondary datapoints and report their own "
                         "dtype")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1
    backend = os.environ.get("MOANN_BENCH_BACKEND", "nccl")
    if distributed:
        import torch.distributed as dist
        dist.init_process_group(backend)
    # modulo lets N-rank validation run on a box with fewer GPUs (the
    # driver's 8-GPU run maps ranks to distinct devices unchanged)
    devno = local_rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(devno)
    dev = torch.device("cuda", devno)

    from matrixone_amd import engine
    from matrixone_amd.distributed import merge_topk_torch

    # ---- data + build (per-rank shard; seed differs per rank) -------------
    t_build0 = time.perf_counter()
    log(f"[bench] generating {args.rows}x{args.dim} f32 mixture shard on GPU ...")
    data, queries_host = make_mixture(args.rows, args.dim, world, rank, dev)
    log("[bench] k-means (harness-side build) ...")
    cent = kmeans_torch(data, args.nlist, iters=8, seed=123 + rank)
    assign = assign_torch(data, cent)
    queries = queries_host.to(dev)
    torch.cuda.synchronize()

    log("[bench] ingest + pack into moann index ...")
    cent_np = cent.cpu().numpy()
    assign_np = assign.cpu().numpy()
    # keep a full host copy only where the CPU baseline will need it
    # (rank 0, N=1); otherwise stream 2M-row chunks through a small host
    # buffer so an 8-rank run never stages 8 x 30 GB in host RAM at once.
    need_host = (rank == 0 and world == 1 and not args.skip_cpu_baseline)
    host_vecs = data.cpu().numpy() if need_host else None
    ix = engine.IvfFlatIndex(args.dim, args.nlist, metric="l2sq",
                             capacity=args.rows, device=devno,
                             qtype=args.qtype)
    CH = 1 << 21
    for c0 in range(0, args.rows, CH):
        if host_vecs is not None:
            ix.add(host_vecs[c0:c0 + CH])
        else:
            ix.add(data[c0:c0 + CH].cpu().numpy())
    ix.set_centroids(cent_np)
    ix.set_assignments(assign_np)
    if args.qtype in ("int8", "uint8"):
        sample = data[:2048].reshape(-1).float()  # quantile caps at 2^24
        lo = float(torch.quantile(sample, 0.001))
        hi = float(torch.quantile(sample, 0.999))
        ix.set_quantizer(lo, hi)
    ix.build()
    if args.refine and args.qtype == "f32":
        try:
            ix.enable_refine(args.refine)
        except Exception as e:  # never let the accelerator mode kill the run
            log(f"[bench] enable_refine failed ({e}); exact one-pass scan")
            args.refine = 0
    else:
        args.refine = 0
    log(f"[bench] build done in {time.perf_counter() - t_build0:.1f}s")

    # ---- recall check ------------------------------------------------------
    nrq = args.recall_queries
    gt_local = exact_gt(data, queries[:nrq], args.k)  # local-shard GT

    def measure_recall():
        ids_r, dists_r = ix.search_device(queries[:nrq].contiguous(),
                                          args.k, args.nprobe)
        if not distributed:
            gt_ids = gt_local.cpu().numpy()
            return float(np.mean([
                len(set(ids_r[i]).intersection(gt_ids[i])) / args.k
                for i in range(nrq)]))
        # merge IVF results across shards (ids are shard-local rows; offset)
        off = rank * args.rows
        ids_off = np.where(ids_r >= 0, ids_r + off, -1)
        from matrixone_amd.distributed import allgather_merge
        m_ids, _ = allgather_merge(ids_off, dists_r, args.k)
        # global GT: merge per-shard exact GT by distance
        gt_dist = gt_exact_dists(data, queries[:nrq], gt_local)
        g_ids, _ = allgather_merge(
            np.where(gt_local.cpu().numpy() >= 0,
                     gt_local.cpu().numpy() + off, -1),
            gt_dist.cpu().numpy().astype(np.float32), args.k)
        return float(np.mean([
            len(set(m_ids[i]).intersection(g_ids[i])) / args.k
            for i in range(nrq)]))

    recall = measure_recall()
    log(f"[bench] recall@{args.k} = {recall:.4f} (target >= 0.95)")
    if recall < 0.95 and args.refine:
        # two-stage scan missed the gate on this data: fall back to the
        # exact one-pass scan and re-measure. recall is identical on every
        # rank (deterministic merged value), so all ranks take this branch
        # together; the reported config is always the one actually benched.
        log("[bench] refine missed the recall gate — falling back to the "
            "exact one-pass scan")
        ix.enable_refine(0)
        args.refine = 0
        recall = measure_recall()
        log(f"[bench] recall@{args.k} = {recall:.4f} (exact path)")

    # ---- timed region ------------------------------------------------------
    nbatches = max(1, queries.shape[0] // args.batch)

    # depth-2 pipelined stepping (moann_ivf_flat_search_submit): batch i+1's
    # centroid rank + host job build overlap batch i's list scan. Disable
    # with MOANN_BENCH_PIPELINE=0 for A/B. Every batch is fully computed and
    # collected inside the timed region.
    pipeline = os.environ.get("MOANN_BENCH_PIPELINE", "1") != "0"

    def submit(i):
        b0 = (i % nbatches) * args.batch
        q = queries[b0:b0 + args.batch].contiguous()
        return ix.search_submit(q, args.k, args.nprobe)

    def merge(ids, dists):
        if distributed:
            from matrixone_amd.distributed import allgather_merge
            off = rank * args.rows
            ids = np.where(ids >= 0, ids + off, -1)
            ids, dists = allgather_merge(ids, dists, args.k)
        return ids

    def step(i):
        b0 = (i % nbatches) * args.batch
        q = queries[b0:b0 + args.batch].contiguous()
        ids, dists = ix.search_device(q, args.k, args.nprobe)
        return merge(ids, dists)

    def run_steps(n):
        if n <= 0:
            return
        if not pipeline:
            for i in range(n):
                step(i)
            return
        t = submit(0)
        for i in range(n):
            nxt = submit(i + 1) if i + 1 < n else None
            ids, dists = ix.search_collect(t)
            merge(ids, dists)
            t = nxt

    run_steps(args.warmup)
    ix.perf_reset()
    if distributed:
        import torch.distributed as dist
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    run_steps(args.steps)
    torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if distributed:
        import torch.distributed as dist
        t = (torch.tensor([elapsed], device=dev)
             if dist.get_backend() == "nccl" else torch.tensor([elapsed]))
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    qps = args.steps * args.batch / elapsed
    perf = ix.perf()
    # Under pipelining, concurrent streams share bandwidth and the event
    # windows stretch past the kernels' standalone durations; re-measure the
    # kernel windows with a short UNPIPELINED probe so the roofline reflects
    # the kernel itself. Step throughput (elapsed) stays pipelined.
    nprobe_steps = 0
    if pipeline:
        ix.perf_reset()
        nprobe_steps = 3
        for i in range(nprobe_steps):
            step(i)
        perf = ix.perf()
        kdiv = nprobe_steps
    else:
        kdiv = args.steps
    log(f"[bench] perf: rank {perf['rank_ms']/kdiv:.3f} ms/step, "
        f"scan {perf['scan_ms']/kdiv:.3f}, "
        f"other {perf.get('other_ms', 0.0)/kdiv:.3f}, "
        f"select {perf['select_ms']/kdiv:.3f}, "
        f"step total {elapsed/args.steps*1e3:.3f} (pipelined={pipeline})")

    # ---- roofline (list-scan kernel, HIP events on the index stream) ------
    # first-pass scan element size: the refine mode's first pass reads the
    # internal byte image regardless of the (f32) storage dtype
    elem = 1 if args.refine else {"f32": 4, "f16": 2,
                                  "bf16": 2}.get(args.qtype, 1)
    scan_gbs = (perf["scan_bytes"] / 4 * elem / 1e9) / max(
        perf["scan_ms"] / 1e3, 1e-9)
    # PMC traffic: per-launch HBM bytes measured with rocprofv3 --pmc
    # FETCH_SIZE against THIS workload+scan mode, committed under profiles/
    # (collected by tools/collect_traffic.sh; x2 gfx950 counter correction
    # per MI355X_MICROARCH.md). Emitted only when the keyed entry matches
    # the exact configuration being run; otherwise null.
    traffic = None
    traffic_src = None
    try:
        tpath = os.path.join(REPO, "profiles", "r02_traffic.json")
        with open(tpath) as f:
            tdb = json.load(f)
        key = (f"{args.rows}x{args.dim}-nlist{args.nlist}-"
               f"nprobe{args.nprobe}-batch{args.batch}-"
               f"{'refine' if args.refine else 'exact'}-{args.qtype}")
        ent = tdb.get(key)
        if ent and world == 1:
            traffic = ent["bytes_per_launch"]
            traffic_src = ent.get("provenance")
    except Exception:
        pass
    roofline = {
        "bound": "hbm",
        "achieved": round(scan_gbs, 1),
        "peak": HBM_PEAK_GBS,
        "unit": "GB/s",
        "frac": round(scan_gbs / HBM_PEAK_GBS, 4),
        "traffic": traffic,
    }
    if traffic_src:
        roofline["traffic_provenance"] = traffic_src

    # ---- CPU baseline (rank 0, N=1 only) -----------------------------------
    cpu_baseline = None
    if host_vecs is not None:
        log("[bench] CPU baseline (oracle port, OpenMP) ...")
        try:
            cpu_baseline = run_cpu_baseline(
                host_vecs, cent_np, assign_np,
                queries_host.numpy(), args.nprobe, args.k,
                args.cpu_baseline_queries)
        except Exception as e:  # report the miss, never fake a number
            log(f"[bench] cpu_baseline failed: {e}")

    if rank == 0:
        line = {
            "metric": "ANN queries/sec @ recall@10>=0.95, 10Mx768 f32",
            "value": round(qps, 1),
            "unit": "queries/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.qtype,
            "data": "synthetic",
            "config": {
                "workload": f"ivfflat-l2-{args.rows}x{args.dim}-"
                            f"nlist{args.nlist}-nprobe{args.nprobe}-"
                            f"batch{args.batch}-k{args.k}",
                "rows_per_gpu": args.rows,
                "dim": args.dim,
                "nlist": args.nlist,
                "nprobe": args.nprobe,
                "global_batch": args.batch,
                "k": args.k,
                "recall_at_k": round(recall, 4),
                "scan": (f"two-stage-refine-R{args.refine}" if args.refine
                         else "exact-one-pass"),
                "parallelism": f"sharded-lists x{world}" if world > 1
                               else "single",
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(line), flush=True)

    ix.close()
    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


def gt_exact_dists(data, queries, gt_ids):
    """Exact squared L2 for each (query, gt id) pair."""
    rows = data[gt_ids.reshape(-1)]
    q = queries.repeat_interleave(gt_ids.shape[1], 0)
    return ((rows - q) ** 2).sum(1).reshape(gt_ids.shape)


exact_gt_dists = gt_exact_dists  # legacy alias


if __name__ == "__main__":
    main()
