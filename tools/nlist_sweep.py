#!/usr/bin/env python3
"""nlist/nprobe sweep on the north-star corpus (VERDICT round-1 item 2).

The round-1 bench measured 3.5x list skew: clustered queries probe the
dense lists, so a query scans ~275k codes instead of the algorithmic
nprobe*N/nlist = 78k. Larger nlist shrinks every list (the reference's
own sizing sweeps, internal/engine/benchs/README.md:33-62); this script
measures QPS + recall@10 across (nlist, nprobe) so the headline config
can be re-tuned while holding the recall@10 >= 0.95 gate.

Recall here is ENGINE-INDEPENDENT: ground truth is an exact fp64
brute-force pass over the full corpus (numpy, host), which also serves
as VERDICT item 8's non-engine truth artifact.

Writes one JSON line per config to stdout and a summary file under
gpurun_out/.
"""
import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench  # noqa: E402  (gen_data, build_engine, query recipe)


def exact_gt_fp64(base, queries, k, chunk=500_000):
    """Exact L2^2 top-k over the whole corpus, fp64 accumulation,
    ties broken by ascending id (the engine's (dist, id) total order).
    Engine-independent: pure numpy on the host."""
    nq = queries.shape[0]
    qd = queries.astype(np.float64)
    best_d = np.full((nq, k), np.inf)
    best_i = np.full((nq, k), -1, dtype=np.int64)
    for s in range(0, base.shape[0], chunk):
        b = base[s:s + chunk].astype(np.float64)
        bn = (b * b).sum(1)
        # ||q-b||^2 computed exactly as sum over d of (q_i-b_i)^2 would
        # order; fp64 dot keeps >52 bits so the order is exact for f32
        # inputs of this scale
        d = (qd * qd).sum(1)[:, None] + bn[None, :] - 2.0 * (qd @ b.T)
        kk = min(k, d.shape[1])
        part = np.argpartition(d, kk - 1, axis=1)[:, :kk]
        cd = np.take_along_axis(d, part, axis=1)
        ci = part + s
        ad = np.concatenate([best_d, cd], axis=1)
        ai = np.concatenate([best_i, ci], axis=1)
        order = np.lexsort((ai, ad), axis=1)[:, :k]
        best_d = np.take_along_axis(ad, order, axis=1)
        best_i = np.take_along_axis(ai, order, axis=1)
    return best_d, best_i


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--db-size", type=int, default=10_000_000)
    ap.add_argument("--nq", type=int, default=10_000)
    ap.add_argument("--nq-gt", type=int, default=512)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--rerank", type=int, default=200)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--nlists", default="4096,8192,16384")
    ap.add_argument("--nprobes", default="24,32,48,64")
    ap.add_argument("--out", default="gpurun_out/nlist_sweep.json")
    args = ap.parse_args()

    d, n = 128, args.db_size
    cfgs = [int(x) for x in args.nlists.split(",")]
    nprobes = [int(x) for x in args.nprobes.split(",")]

    t0 = time.time()
    base = bench.gen_data(n, d, seed=42)
    rng = np.random.default_rng(43)
    idx = rng.integers(0, n, size=args.nq)
    queries = base[idx] + 0.05 * rng.standard_normal(
        (args.nq, d), dtype=np.float32)
    print(f"[sweep] data gen {time.time()-t0:.1f}s", file=sys.stderr,
          flush=True)

    t0 = time.time()
    gt_d, gt_i = exact_gt_fp64(base, queries[:args.nq_gt], args.k)
    print(f"[sweep] fp64 exact GT over {n} docs x {args.nq_gt} queries "
          f"in {time.time()-t0:.1f}s", file=sys.stderr, flush=True)

    results = []
    for nlist in cfgs:
        cfg = dict(kind="IVFPQ", d=d, n=n, nlist=nlist, m=32, nprobe=32,
                   nq=args.nq, k=args.k, rerank=args.rerank,
                   train_n=min(n, max(160_000, 39 * nlist)))
        t0 = time.time()
        eng = bench.build_engine(cfg, base, 0,
                                 path=f"/tmp/gamma_sweep_{nlist}")
        t_build = time.time() - t0
        nq = eng.cache_queries(queries)
        for nprobe in nprobes:
            # recall vs fp64 GT
            gd, gi = eng.raw_search(queries[:args.nq_gt], args.k,
                                    nprobe=nprobe, rerank=args.rerank)
            hits = sum(
                len(set(gt_i[t].tolist()) & set(int(x) for x in gi[t]
                                                if x >= 0))
                for t in range(args.nq_gt))
            recall = hits / (args.nq_gt * args.k)
            # scanned codes/query (skew factor)
            eff = None
            try:
                from vearch_amd.engine import lib as _lib
                _, pl = eng.debug_coarse_assign(queries[:256], nprobe)
                sizes = {int(ln): _lib().GammaDebugGetList(
                    eng.h, int(ln), None, None)
                    for ln in np.unique(pl) if ln >= 0}
                eff = float(np.mean(
                    [sum(sizes.get(int(x), 0) for x in row)
                     for row in pl]))
            except Exception:
                pass
            # QPS
            for _ in range(args.warmup):
                eng.search_cached(nq, args.k, nprobe=nprobe,
                                  rerank=args.rerank)
            t0 = time.time()
            for _ in range(args.steps):
                eng.search_cached(nq, args.k, nprobe=nprobe,
                                  rerank=args.rerank)
            dt = time.time() - t0
            line = {
                "nlist": nlist,
                "nprobe": nprobe,
                "rerank": args.rerank,
                "qps": round(args.nq * args.steps / dt, 1),
                "ms_per_step": round(dt / args.steps * 1e3, 3),
                "recall_at_10_vs_fp64_gt": round(recall, 4),
                "scanned_codes_per_query": eff,
                "algorithmic_codes_per_query": nprobe * n / nlist,
                "train_n": cfg["train_n"],
                "build_s": round(t_build, 1),
                "stage_us": eng.last_timing(),
            }
            results.append(line)
            print(json.dumps(line), flush=True)
        eng.close()

    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, "w") as f:
        json.dump({"db_size": n, "nq": args.nq, "nq_gt": args.nq_gt,
                   "k": args.k, "note": "recall vs engine-independent "
                   "fp64 exact brute force (numpy host)",
                   "results": results}, f, indent=1)
    print(f"[sweep] wrote {args.out}", file=sys.stderr)


if __name__ == "__main__":
    main()
