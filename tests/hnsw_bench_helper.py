"""HNSW GPU-vs-usearch measurement helper (run on a GPU box; invoked by hand
or from gpurun — not part of the pytest suite).

Builds a usearch graph (the reference engine, multi-threaded), exports it,
ingests into the MI355X kernel, and reports:
  - GPU QPS at the config-3 operating point shape (cosine, M=32, ef=128,
    batch=256) scaled to --rows,
  - usearch CPU QPS on the same graph and queries (the reference engine,
    kind "reference" baseline),
  - agreement between the two result sets.
"""

import argparse
import json
import sys
import time

import numpy as np

sys.path.insert(0, ".")

from oracle import hnsw as H  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=200_000)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--conn", type=int, default=32)
    ap.add_argument("--ef", type=int, default=128)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--threads", type=int, default=64)
    ap.add_argument("--cpu-queries", type=int, default=512)
    args = ap.parse_args()

    rng = np.random.Generator(np.random.PCG64(4242))
    latent = 32
    U = rng.standard_normal((latent, args.dim), dtype=np.float32) / np.sqrt(latent)
    ncomp = max(256, args.rows // 100)
    z = rng.standard_normal((ncomp, latent), dtype=np.float32) * np.sqrt(latent)
    centers = z @ U
    comp = rng.integers(0, ncomp, args.rows)
    vecs = centers[comp] + rng.standard_normal((args.rows, args.dim),
                                               dtype=np.float32)
    qcomp = rng.integers(0, ncomp, 4096)
    queries = centers[qcomp] + rng.standard_normal((4096, args.dim),
                                                   dtype=np.float32)

    t0 = time.perf_counter()
    ref = H.RefHnsw(args.dim, metric=H.METRIC_COS, connectivity=args.conn,
                    expansion_add=128, expansion_search=args.ef,
                    capacity=args.rows, threads=args.threads)
    ref.add(vecs)
    t_build = time.perf_counter() - t0
    t0 = time.perf_counter()
    g = ref.export_graph()
    t_export = time.perf_counter() - t0

    from matrixone_amd.hnsw import HnswGraphData, HnswIndex
    gd = HnswGraphData(dim=g.dim, metric="cos", entry_slot=g.entry_slot,
                       max_level=g.max_level, levels=g.levels, keys=g.keys,
                       vecs=g.vecs, adj=g.adj)
    gpu = HnswIndex(gd)

    # warmup + timed GPU steps
    gpu.search(queries[:args.batch], args.ef, args.k)
    gpu.perf_reset()
    t0 = time.perf_counter()
    for i in range(args.steps):
        b0 = (i * args.batch) % (4096 - args.batch)
        gpu.search(queries[b0:b0 + args.batch], args.ef, args.k)
    gpu_dt = time.perf_counter() - t0
    perf = gpu.perf()

    # usearch CPU on the same graph (multi-thread via python loop is serial;
    # time the C engine serially and scale by threads is dishonest — report
    # SINGLE-thread usearch QPS and the thread count separately)
    nq = args.cpu_queries
    t0 = time.perf_counter()
    ref_ids, _ = ref.search(queries[:nq], ef=args.ef, k=args.k)
    cpu_dt = time.perf_counter() - t0

    gids, _ = gpu.search(queries[:nq], args.ef, args.k)
    agree = float((gids == ref_ids).mean())

    out = {
        "workload": f"hnsw-cos-{args.rows}x{args.dim}-M{args.conn}-"
                    f"ef{args.ef}-batch{args.batch}-k{args.k}",
        "gpu_qps": round(args.steps * args.batch / gpu_dt, 1),
        "gpu_ms_per_batch": round(gpu_dt / args.steps * 1e3, 3),
        "gpu_dist_evals_per_query": round(
            perf["scan_rows"] / max(1, perf["scan_launches"] * args.batch), 1),
        "gpu_eval_GBps": round(perf["scan_bytes"] / 1e9
                               / max(perf["scan_ms"] / 1e3, 1e-9), 1),
        "usearch_cpu_qps_1thread": round(nq / cpu_dt, 1),
        "gpu_vs_usearch_id_agreement": round(agree, 4),
        "build_s": round(t_build, 1),
        "export_s": round(t_export, 1),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
