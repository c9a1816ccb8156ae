"""Small-batch latency probe (the serving path; probe-split S kicks in
below nq=1024 WGs). Run on a GPU box: python tools/latency_probe.py"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import oracle as orc  # noqa: E402
from vearch_amd import GammaEngine  # noqa: E402


def main():
    n = 2_000_000
    base = orc.gen_clustered(n, 128, seed=42, ncl=10000)
    eng = GammaEngine(path="/tmp/gamma_lat")
    eng.create_table(
        128, "IVFPQ",
        '{"ncentroids": 4096, "nsubvector": 32, "metric_type": "L2", '
        '"training_threshold": 160000}')
    eng.add(base)
    eng.build_index()
    for nq in (1, 16, 256, 2048):
        q = orc.gen_queries(base, nq, seed=5)
        eng.raw_search(q, 10, nprobe=32, rerank=200)  # warm
        ts = []
        for _ in range(20):
            t0 = time.time()
            eng.raw_search(q, 10, nprobe=32, rerank=200)
            ts.append((time.time() - t0) * 1e3)
        ts.sort()
        print(f"nq={nq:5d}  p50={ts[10]:7.3f} ms  p90={ts[18]:7.3f} ms  "
              f"QPS={nq / (ts[10] / 1e3):9.0f}")
    eng.close()


if __name__ == "__main__":
    main()
