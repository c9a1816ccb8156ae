#!/usr/bin/env python3
"""Throughput vs batch size on the headline index (10M docs,
nlist=16384, nprobe=32, recall_num=200): the serving-side counterpart
of the nq=10k bench line. Run on a GPU box with a prepared index:

    python bench.py --index-dir /tmp/idx --prepare-only
    python tools/qps_curve.py --index-dir /tmp/idx
"""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--index-dir", default="/tmp/idx")
    args = ap.parse_args()
    cfg = dict(bench.WORKLOADS["ivfpq_d128_n10m_nprobe32"])
    from vearch_amd import GammaEngine
    eng = GammaEngine(path=args.index_dir)
    params = ('{"ncentroids": %d, "nsubvector": %d, "metric_type": "L2", '
              '"training_threshold": %d, "bucket_max_size": 12800000}'
              % (cfg["nlist"], cfg["m"], cfg["train_n"]))
    eng.create_table(cfg["d"], "IVFPQ", params)
    t0 = time.time()
    eng.load()
    print(f"[qps_curve] loaded {eng.num_docs()} docs in "
          f"{time.time()-t0:.1f}s", file=sys.stderr)
    queries = np.load(os.path.join(args.index_dir, "queries.npy"))
    for nq in (1, 16, 64, 256, 1024, 4096, 10000):
        q = queries[:nq]
        lat = []
        reps = max(3, min(50, 2000 // max(nq // 16, 1)))
        for _ in range(2):
            eng.raw_search(q, cfg["k"], nprobe=cfg["nprobe"],
                           rerank=cfg["rerank"])
        for _ in range(reps):
            t0 = time.time()
            eng.raw_search(q, cfg["k"], nprobe=cfg["nprobe"],
                           rerank=cfg["rerank"])
            lat.append(time.time() - t0)
        lat.sort()
        p50 = lat[len(lat) // 2] * 1e3
        p90 = lat[int(len(lat) * 0.9)] * 1e3
        print(f"nq={nq:6d}  p50={p50:8.3f} ms  p90={p90:8.3f} ms  "
              f"QPS={nq/ (p50/1e3):12.1f}", flush=True)
    eng.close()


if __name__ == "__main__":
    main()
