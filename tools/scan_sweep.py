#!/usr/bin/env python3
"""Scan-kernel launch-shape sweep (round-2 perf): probe-split S x block
size BS on the north-star workload. The scan at nlist=16384 runs short
per-(query,list) bursts (~N/nlist codes against 512 threads); S>1
spreads a query's probes over sub-workgroups and BS=256 halves the
tail waste. Exactness is unchanged by construction (partials merge in
the row sort; parity suite runs S>1 at small nq) and double-checked
here against the S=1/BS=512 ids.

Usage (GPU box): python tools/scan_sweep.py --index-dir /tmp/idx
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
    ap.add_argument("--steps", type=int, default=5)
    args = ap.parse_args()

    cfg = dict(bench.WORKLOADS["ivfpq_d128_n10m_nprobe32"])
    from vearch_amd import GammaEngine
    eng = GammaEngine(path=args.index_dir)
    params = ('{"ncentroids": %d, "nsubvector": %d, "metric_type": "L2", '
              '"training_threshold": %d, "bucket_max_size": 12800000}'
              % (cfg["nlist"], cfg["m"], cfg["train_n"]))
    eng.create_table(cfg["d"], "IVFPQ", params)
    eng.load()
    queries = np.load(os.path.join(args.index_dir, "queries.npy"))
    nq = eng.cache_queries(queries)
    print(f"[scan_sweep] loaded {eng.num_docs()} docs", file=sys.stderr)

    ref_ids = None
    for bs in (512, 256):
        for s_split in (1, 2, 4):
            os.environ["GAMMA_SCAN_BS"] = str(bs)
            os.environ["GAMMA_SCAN_S"] = str(s_split)
            for _ in range(2):
                eng.search_cached(nq, cfg["k"], nprobe=cfg["nprobe"],
                                  rerank=cfg["rerank"])
            t0 = time.time()
            scan_us = []
            for _ in range(args.steps):
                d_, i_ = eng.search_cached(nq, cfg["k"],
                                           nprobe=cfg["nprobe"],
                                           rerank=cfg["rerank"])
                scan_us.append(eng.last_timing()["scan_us"])
            dt = time.time() - t0
            if ref_ids is None:
                ref_ids = i_.copy()
                same = True
            else:
                same = bool(np.array_equal(i_, ref_ids))
            print(f"BS={bs} S={s_split}: scan {np.mean(scan_us):8.1f} us "
                  f"step {dt/args.steps*1e3:7.3f} ms "
                  f"qps {nq*args.steps/dt:10.1f} ids_equal={same}",
                  flush=True)
    eng.close()


if __name__ == "__main__":
    main()
