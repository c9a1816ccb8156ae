"""Generate committed golden fixtures for the oracle (tests/golden/).

Run from the repo root: python tests/gen_golden.py
Deterministic (fixed seeds). The .npz outputs are committed so later
refactors of the oracle/engine are regression-pinned to these vectors.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oracle import (OracleIVFPQ, flat_search, flat_topk_f64, gen_clustered,
                    gen_queries, kmeans, pq_train)
from oracle.gamma_oracle import RefLib, _fp, _up8, _c

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def main():
    os.makedirs(GOLDEN, exist_ok=True)
    d, n, nq, k = 32, 2000, 16, 5

    base = gen_clustered(n, d, seed=7, ncl=50)
    q = gen_queries(base, nq, seed=8)

    # FLAT L2 + IP
    fd, fi = flat_search(base, q, k, "L2")
    fdip, fiip = flat_search(base, q, k, "InnerProduct")
    gtd, gti = flat_topk_f64(base, q, k, "L2")

    # FLAT with deletes: delete every 7th vid
    bm = np.zeros((n + 7) // 8, dtype=np.uint8)
    for vid in range(0, n, 7):
        bm[vid >> 3] |= 1 << (vid & 7)
    fdd, fid = flat_search(base, q, k, "L2", del_bitmap=bm)

    # ADC tables for one fixed query/centroid/codebook
    M, ksub, dsub = 8, 256, d // 8
    rng = np.random.default_rng(11)
    cent1 = rng.random(d, dtype=np.float32)
    books = rng.standard_normal((M, ksub, dsub)).astype(np.float32) * 0.1
    tab_l2 = np.empty((M, ksub), dtype=np.float32)
    tab_ip = np.empty((M, ksub), dtype=np.float32)
    lib = RefLib.lib()
    q0 = _c(q[0], np.float32)
    lib.oracle_adc_table_l2(d, M, ksub, _fp(q0), _fp(_c(cent1, np.float32)),
                            _fp(_c(books, np.float32)), _fp(tab_l2))
    lib.oracle_adc_table_ip(d, M, ksub, _fp(q0),
                            _fp(_c(books, np.float32)), _fp(tab_ip))

    # PQ encode golden
    resid = rng.standard_normal((64, d)).astype(np.float32) * 0.1
    codes = np.empty((64, M), dtype=np.uint8)
    lib.oracle_pq_encode(64, d, M, ksub, _fp(_c(resid, np.float32)),
                         _fp(_c(books, np.float32)), _up8(codes))

    # k-means determinism golden (small)
    cent_km = kmeans(base[:800], 16, niter=10, seed=42)

    # IVFPQ end-to-end small
    ix = OracleIVFPQ(d, 32, M)
    ix.train(base[:1500], seed=42)
    ix.add(base)
    pd_, pl = ix.coarse_assign(q, 8)
    sd, si = ix.search(q, k, nprobe=8)

    np.savez_compressed(
        os.path.join(GOLDEN, "oracle_golden.npz"),
        base_sum=np.float64(base.astype(np.float64).sum()),
        flat_l2_d=fd, flat_l2_i=fi, flat_ip_d=fdip, flat_ip_i=fiip,
        flat_f64_i=gti,
        flat_del_d=fdd, flat_del_i=fid, del_bitmap=bm,
        adc_tab_l2=tab_l2, adc_tab_ip=tab_ip,
        pq_codes=codes,
        kmeans_cent=cent_km,
        ivfpq_centroids=ix.centroids, ivfpq_codebooks=ix.codebooks,
        ivfpq_probe_d=pd_, ivfpq_probe_l=pl,
        ivfpq_d=sd, ivfpq_i=si,
    )
    print("written", os.path.join(GOLDEN, "oracle_golden.npz"))


if __name__ == "__main__":
    main()
