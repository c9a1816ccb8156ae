"""Filtered-search timing probe (SURVEY §8f-2 perf evidence; round-2
plan item 4). Uses only the validated product path: docs with a scalar
tag field, protobuf Search with a term filter at several selectivities,
wall-clock per batch. Run on a GPU box:

    python tools/filter_probe.py
"""
import os
import struct
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import oracle as orc  # noqa: E402
from vearch_amd import GammaEngine, fbsenc  # noqa: E402


def main():
    n, d = 400_000, 64
    base = orc.gen_clustered(n, d, seed=42, ncl=2000)
    eng = GammaEngine(path="/tmp/gamma_filterprobe")
    eng.create_table(
        d, "IVFPQ",
        '{"ncentroids": 512, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 80000}',
        scalar_fields=[("tag", fbsenc.DATA_INT)])
    t0 = time.time()
    for vid in range(n):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", struct.pack("<i", vid % 1000),
                             fbsenc.DATA_INT)])
    print(f"ingest {n} docs with tags: {time.time() - t0:.1f}s")
    eng.build_index()
    q = orc.gen_queries(base, 64, seed=5)

    def timed(label, **kw):
        eng.search_pb(q, topn=10, **kw)  # warm (also builds the index)
        ts = []
        for _ in range(10):
            t = time.time()
            eng.search_pb(q, topn=10, **kw)
            ts.append((time.time() - t) * 1e3)
        ts.sort()
        print(f"{label:42s} p50={ts[5]:7.2f} ms/64q "
              f"({64 / (ts[5] / 1e3):8.0f} QPS)")

    timed("unfiltered")
    # term filter: 1/1000 of docs match
    timed("term filter (0.1% selectivity)",
          term_filters=[("tag", b"\x07\x00\x00\x00")])
    # range filters at three selectivities
    for lo, hi, lbl in ((0, 9, "1%"), (0, 99, "10%"), (0, 499, "50%")):
        timed(f"range filter ({lbl} selectivity)",
              range_filters=[("tag", struct.pack("<i", lo),
                              struct.pack("<i", hi), True, True)])
    eng.close()


if __name__ == "__main__":
    main()
