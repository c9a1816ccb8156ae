"""Generate tests/golden/ref_golden.npz — outputs of the COMPILED
REFERENCE scanner (oracle/_ref/libgammaref.so, the reference's own
extracted code; see oracle/ref_harness.cpp) on seeded inputs, together
with the exact model/list arrays they were produced from.

Run from the repo root (requires /root/reference to have been present
at build time): python tests/gen_ref_golden.py

tests/test_ref_pin.py then pins oracle/ref_scan.c bit-exactly against
these reference-produced vectors ANYWHERE (no reference, no _ref lib
needed at test time) — this is what turns "the oracle restates the
reference" from a reading claim into a tested one.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import oracle as orc  # noqa: E402
from oracle import refbind as rb  # noqa: E402
from oracle.gamma_oracle import RefLib, _c, _fp, _ip64  # noqa: E402

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def main():
    assert rb.available(), "oracle/_ref/libgammaref.so not built"
    os.makedirs(GOLDEN, exist_ok=True)
    out = {}

    # ---- FLAT (gamma_index_flat.cc:48-130 compiled) ----
    d, n, nq, k = 32, 5000, 16, 10
    base = orc.gen_clustered(n, d, seed=7, ncl=50)
    q = orc.gen_queries(base, nq, seed=8)
    bm = np.zeros((n + 7) // 8, np.uint8)
    for vid in range(0, n, 7):
        bm[vid >> 3] |= 1 << (vid & 7)
    out["flat_base_seed"] = np.int64(7)
    out["flat_base"] = base
    out["flat_q"] = q
    out["flat_bm"] = bm
    for metric in ("L2", "IP"):
        rd, ri = rb.flat_search(base, q, k, metric)
        out[f"flat_{metric}_d"], out[f"flat_{metric}_i"] = rd, ri
        rdd, rid = rb.flat_search(base, q, k, metric, del_bitmap=bm)
        out[f"flat_{metric}_del_d"] = rdd
        out[f"flat_{metric}_del_i"] = rid

    # ---- IVFPQ (gamma_index_ivfpq.h:76-389 + 923-953 compiled) ----
    d, nlist, M, n, nq, k, nprobe = 64, 32, 16, 8000, 16, 10, 8
    base = orc.gen_clustered(n, d, seed=11, ncl=40)
    q = orc.gen_queries(base, nq, seed=12)
    ox = orc.OracleIVFPQ(d, nlist, M, metric="L2")
    ox.train(base[:4000])
    ox.add(base)
    # deletes via both channels: bit 63 in list ids + the IsValid bitmap
    ox.ids[::11] |= np.int64(np.uint64(1) << np.uint64(63))
    pbm = np.zeros((n + 7) // 8, np.uint8)
    for vid in range(0, n, 13):
        pbm[vid >> 3] |= 1 << (vid & 7)
    _, probes = ox.coarse_assign(q, nprobe)
    gdists = ox.coarse_gemm_dists(q, probes)
    out["pq_centroids"] = ox.centroids
    out["pq_codebooks"] = ox.codebooks
    out["pq_offsets"] = ox.offsets
    out["pq_ids"] = ox.ids
    out["pq_codes"] = ox.codes
    out["pq_q"] = q
    out["pq_probes"] = probes
    out["pq_gdists"] = gdists
    out["pq_bm"] = pbm
    for upt in (0, 1):
        rd, ri = rb.ivfpq_search(ox, q, k, nprobe, probes, gdists, upt,
                                 del_bitmap=pbm)
        out[f"pq_upt{upt}_d"], out[f"pq_upt{upt}_i"] = rd, ri
    # IP metric (init_query_IP + precompute_list_tables_IP path)
    oxip = orc.OracleIVFPQ(d, nlist, M, metric="IP")
    oxip.centroids, oxip.codebooks = ox.centroids, ox.codebooks
    oxip.offsets, oxip.ids, oxip.codes = ox.offsets, ox.ids, ox.codes
    rdip, riip = rb.ivfpq_search(oxip, q, k, nprobe, probes, gdists, 0,
                                 metric="IP")
    out["pq_ip_d"], out["pq_ip_i"] = rdip, riip

    # ---- IVFFLAT (gamma_index_ivfflat.h:35-92 compiled) ----
    d, nlist, n, nq, k, nprobe = 32, 24, 4000, 12, 10, 6
    base = orc.gen_clustered(n, d, seed=21, ncl=30)
    q = orc.gen_queries(base, nq, seed=22)
    cent = orc.kmeans(base[:2000], nlist, niter=10, seed=42)
    lib = RefLib.lib()
    pa = np.empty((n, 1), np.int64)
    da = np.empty((n, 1), np.float32)
    lib.oracle_coarse_assign(n, d, nlist, _fp(_c(base, np.float32)),
                             _fp(_c(cent, np.float32)), 1, 0, _fp(da),
                             _ip64(pa))
    asg = pa[:, 0]
    order = np.argsort(asg, kind="stable")
    ids = order.astype(np.int64)
    ids[::5] |= np.int64(np.uint64(1) << np.uint64(63))
    vecs = np.ascontiguousarray(base[order])
    offsets = np.concatenate(
        [[0], np.cumsum(np.bincount(asg, minlength=nlist))]).astype(np.int64)
    probes = np.empty((nq, nprobe), np.int64)
    qd = np.empty((nq, nprobe), np.float32)
    lib.oracle_coarse_assign(nq, d, nlist, _fp(_c(q, np.float32)),
                             _fp(_c(cent, np.float32)), nprobe, 0, _fp(qd),
                             _ip64(probes))
    out["ivff_offsets"] = offsets
    out["ivff_ids"] = ids
    out["ivff_vecs"] = vecs
    out["ivff_q"] = q
    out["ivff_probes"] = probes
    for metric in ("L2", "IP"):
        rd, ri = rb.ivfflat_search(d, nlist, offsets, ids, vecs, q, k,
                                   nprobe, probes, metric)
        out[f"ivff_{metric}_d"], out[f"ivff_{metric}_i"] = rd, ri

    path = os.path.join(GOLDEN, "ref_golden.npz")
    np.savez_compressed(path, **out)
    print(f"wrote {path} ({os.path.getsize(path)/1e6:.2f} MB)")


if __name__ == "__main__":
    main()
