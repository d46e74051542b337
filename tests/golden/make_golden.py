"""Generate the committed golden fixtures in tests/golden/.

Inputs are seeded (PCG64) and stored alongside outputs, so the fixtures pin
BOTH the oracle restatement and — where /root/reference is present — the
reference's own arithmetic (usearch_exact_search over the same arrays).

Run from the repo root:  python tests/golden/make_golden.py
Commit the produced .npz files. Keep cases small (< 1 MB total).
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

from oracle import oracle as orc  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))


def gen_case_ivf(name, n, d, nlist, nq, probe, k, metric, orig_l2, seed):
    rng = np.random.Generator(np.random.PCG64(seed))
    vecs = rng.standard_normal((n, d), dtype=np.float32)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    # centroids: k-means-lite (sampled init + 5 Lloyd iterations in f64 numpy —
    # build-time machinery, shared verbatim by oracle and product via the
    # stored arrays; parity never depends on this converging well)
    cidx = rng.choice(n, size=nlist, replace=False)
    cent = vecs[cidx].astype(np.float64)
    for _ in range(5):
        d2 = ((vecs[:, None, :].astype(np.float64) - cent[None, :, :]) ** 2).sum(-1)
        assign = d2.argmin(1)
        for c in range(nlist):
            m = assign == c
            if m.any():
                cent[c] = vecs[m].mean(0)
    cent = cent.astype(np.float32)
    d2 = ((vecs[:, None, :].astype(np.float64) - cent[None, :, :].astype(np.float64)) ** 2).sum(-1)
    assign = d2.argmin(1).astype(np.int64)
    ids = rng.permutation(np.arange(1000, 1000 + n)).astype(np.int64)

    idx = orc.IvfIndex(cent, vecs, assign, ids=ids)
    out_ids, out_dists, out_probe = idx.search(
        metric, queries, probe, k, orig_l2=orig_l2, want_probe=True)
    np.savez_compressed(
        os.path.join(HERE, f"{name}.npz"),
        vecs=vecs, queries=queries, centroids=cent, assign=assign, ids=ids,
        metric=np.int64(metric), orig_l2=np.int64(orig_l2),
        probe=np.int64(probe), k=np.int64(k),
        out_ids=out_ids, out_dists=out_dists, out_probe=out_probe)
    print(f"{name}: n={n} d={d} nlist={nlist} probe={probe} k={k} metric={metric}")


def gen_case_bf(name, n, d, nq, k, metric, orig_l2, seed):
    rng = np.random.Generator(np.random.PCG64(seed))
    base = rng.standard_normal((n, d), dtype=np.float32)
    queries = rng.standard_normal((nq, d), dtype=np.float32)
    out_ids, out_dists = orc.bruteforce(metric, base, queries, k, orig_l2=orig_l2)
    payload = dict(base=base, queries=queries, metric=np.int64(metric),
                   orig_l2=np.int64(orig_l2), k=np.int64(k),
                   out_ids=out_ids, out_dists=out_dists)
    # Pin against the reference's own arithmetic where buildable.
    us_metric = {orc.METRIC_L2SQ: "l2sq", orc.METRIC_IP: "ip",
                 orc.METRIC_COS: "cos"}.get(metric)
    if us_metric is not None and orc.usearch_lib() is not None:
        uk, ud = orc.usearch_exact_search(base, queries, k, metric=us_metric)
        payload["usearch_keys"] = uk
        payload["usearch_dists"] = ud
    np.savez_compressed(os.path.join(HERE, f"{name}.npz"), **payload)
    print(f"{name}: n={n} d={d} k={k} metric={metric} "
          f"(usearch pinned: {'usearch_keys' in payload})")


def main():
    # IVF cases — small versions of BASELINE.json config 1 shapes
    gen_case_ivf("ivf_l2", 4000, 32, 32, 16, 4, 10, orc.METRIC_L2SQ, False, 42)
    gen_case_ivf("ivf_l2_sqrt", 4000, 32, 32, 16, 4, 10, orc.METRIC_L2, True, 43)
    gen_case_ivf("ivf_ip", 4000, 32, 32, 16, 4, 10, orc.METRIC_IP, False, 44)
    gen_case_ivf("ivf_cos", 4000, 32, 32, 16, 4, 10, orc.METRIC_COS, False, 45)
    # ragged dim / edge shapes
    gen_case_ivf("ivf_d33", 1000, 33, 8, 8, 3, 7, orc.METRIC_L2SQ, False, 46)
    gen_case_ivf("ivf_kbig", 300, 16, 4, 4, 4, 64, orc.METRIC_L2SQ, False, 47)
    # brute force
    gen_case_bf("bf_l2", 2000, 64, 16, 10, orc.METRIC_L2SQ, False, 48)
    gen_case_bf("bf_ip", 2000, 64, 16, 10, orc.METRIC_IP, False, 49)
    gen_case_bf("bf_cos", 2000, 64, 16, 10, orc.METRIC_COS, False, 50)


if __name__ == "__main__":
    main()


def make_docfilter_golden():
    """Golden membership-filter payloads + expected membership, built by the
    REFERENCE's own filter code (oracle/_ref/libref_filters.so — cgo/bloom.c,
    cbitmap.c, sorted64.c, croaring.c + vendored CRoaring). The payloads
    travel as fixtures so the GPU box (no /root/reference) can still pin
    matrixone_amd's independent parsers bit-exactly."""
    from oracle import docfilter as DF
    rng = np.random.Generator(np.random.PCG64(20260915))
    keys = rng.choice(1_000_000, 20000, replace=False).astype(np.int64)
    probe = np.concatenate([
        rng.choice(1_200_000, 20000, replace=False),
        keys[:1000],
        [0, 1, 999_999, 1_199_999],
    ]).astype(np.int64)
    out = {}
    pay = DF.build_bloom(keys, nbits=1 << 18, k=5, seed=0x5EED_0F_B100)
    out["bloom_payload"] = np.frombuffer(pay, dtype=np.uint8)
    out["bloom_expect"] = DF.ref_bloom_test(pay, probe)
    pc = DF.build_cbitmap(keys)
    out["cbitmap_payload"] = np.frombuffer(pc, dtype=np.uint8)
    out["cbitmap_expect"] = DF.ref_contains(DF.KIND_CBITMAP, pc, probe)
    ps = DF.build_sorted64(keys)
    out["sorted64_payload"] = np.frombuffer(ps, dtype=np.uint8)
    out["sorted64_expect"] = DF.ref_contains(DF.KIND_SORTED64, ps, probe)
    # croaring: mixed container shapes in one bitmap (array + bitset + runs
    # + a high 32-bit bucket)
    ck = np.concatenate([
        keys[:4000],
        np.arange(2_000_000, 2_070_000),          # bitset container
        np.arange(5_000_000, 5_004_000),          # run container
        rng.choice(10_000, 2000, replace=False) + (1 << 33),
    ]).astype(np.int64)
    pr = DF.build_croaring(ck)
    cprobe = np.concatenate([probe, ck[::5], ck + 1,
                             rng.integers(0, 2**34, 4000)]).astype(np.int64)
    out["croaring_payload"] = np.frombuffer(pr, dtype=np.uint8)
    out["croaring_expect"] = DF.ref_contains(DF.KIND_CROARING, pr, cprobe)
    out["probe_keys"] = probe
    out["croaring_probe_keys"] = cprobe
    np.savez_compressed(os.path.join(HERE, "docfilter.npz"), **out)
    print("docfilter.npz written")


if __name__ == "__main__" and os.environ.get("MOANN_GOLDEN_DOCFILTER"):
    make_docfilter_golden()
