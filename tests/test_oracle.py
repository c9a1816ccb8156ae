"""Oracle self-tests: golden-vector regression pins + the reference's own
recall gates restated:
  - FLAT exactness: test/test_vector_index_flat.py:95-96 (R@1>=0.95,
    R@10>=1.0 -> FLAT is exact, so we assert identity vs fp64 truth).
  - IVFPQ floors: test/test_vector_index_ivfpq.py:106-111.
  - realtime delete semantics: bit-63 mask + bitmap
    (realtime_mem_data.h:26, gamma_index_ivfpq.h:930-935).
"""
import os

import numpy as np
import pytest

from oracle import (OracleIVFPQ, flat_search, flat_topk_f64, gen_clustered,
                    gen_queries, kmeans, recall_at)
from oracle.gamma_oracle import RefLib, _fp, _up8, _c, _ip64

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden",
                      "oracle_golden.npz")


@pytest.fixture(scope="module")
def golden():
    return np.load(GOLDEN)


@pytest.fixture(scope="module")
def small():
    base = gen_clustered(2000, 32, seed=7, ncl=50)
    q = gen_queries(base, 16, seed=8)
    return base, q


def test_flat_golden(golden, small):
    base, q = small
    fd, fi = flat_search(base, q, 5, "L2")
    assert np.array_equal(fi, golden["flat_l2_i"])
    assert np.array_equal(fd, golden["flat_l2_d"])
    fdip, fiip = flat_search(base, q, 5, "InnerProduct")
    assert np.array_equal(fiip, golden["flat_ip_i"])
    assert np.array_equal(fdip, golden["flat_ip_d"])


def test_flat_exact_vs_f64(golden):
    # FLAT is exact: fp32 canonical ids == fp64 ids on this data
    assert np.array_equal(golden["flat_l2_i"], golden["flat_f64_i"])


def test_flat_delete_bitmap(golden, small):
    base, q = small
    bm = golden["del_bitmap"]
    fdd, fid = flat_search(base, q, 5, "L2", del_bitmap=bm)
    assert np.array_equal(fid, golden["flat_del_i"])
    # no deleted vid ever returned
    assert not any(v % 7 == 0 for v in fid.ravel().tolist() if v >= 0)


def test_adc_tables_golden(golden, small):
    base, q = small
    d, M, ksub = 32, 8, 256
    rng = np.random.default_rng(11)
    cent1 = rng.random(d, dtype=np.float32)
    books = rng.standard_normal((M, ksub, d // M)).astype(np.float32) * 0.1
    tab = np.empty((M, ksub), dtype=np.float32)
    lib = RefLib.lib()
    q0 = _c(q[0], np.float32)
    lib.oracle_adc_table_l2(d, M, ksub, _fp(q0), _fp(_c(cent1, np.float32)),
                            _fp(_c(books, np.float32)), _fp(tab))
    assert np.array_equal(tab, golden["adc_tab_l2"])
    # semantic check vs numpy fp64: T[m][j] ~ ||r_m - cw||^2
    r = (q[0] - cent1).astype(np.float64)
    m, j = 3, 77
    want = ((r[m * 4:(m + 1) * 4] - books[m, j].astype(np.float64)) ** 2).sum()
    assert abs(tab[m, j] - want) < 1e-5
    lib.oracle_adc_table_ip(d, M, ksub, _fp(q0),
                            _fp(_c(books, np.float32)), _fp(tab))
    assert np.array_equal(tab, golden["adc_tab_ip"])


def test_pq_encode_golden(golden):
    d, M, ksub = 32, 8, 256
    rng = np.random.default_rng(11)
    _ = rng.random(d, dtype=np.float32)
    books = rng.standard_normal((M, ksub, d // M)).astype(np.float32) * 0.1
    resid = rng.standard_normal((64, d)).astype(np.float32) * 0.1
    codes = np.empty((64, M), dtype=np.uint8)
    RefLib.lib().oracle_pq_encode(64, d, M, ksub,
                                  _fp(_c(resid, np.float32)),
                                  _fp(_c(books, np.float32)), _up8(codes))
    assert np.array_equal(codes, golden["pq_codes"])
    # argmin semantics vs numpy
    dsub = d // M
    for i in (0, 13):
        for m in (0, 5):
            dd = ((resid[i, m * dsub:(m + 1) * dsub][None, :]
                   - books[m]) ** 2).sum(1)
            assert codes[i, m] == np.argmin(dd)


def test_kmeans_golden_and_quality(golden, small):
    base, _ = small
    cent = kmeans(base[:800], 16, niter=10, seed=42)
    assert np.array_equal(cent, golden["kmeans_cent"])
    # quality: k-means beats random centers on quantization error
    rng = np.random.default_rng(0)
    rand_cent = base[rng.choice(800, 16, replace=False)]

    def qerr(c):
        d2 = ((base[:800, None, :].astype(np.float64)
               - c[None, :, :].astype(np.float64)) ** 2).sum(2)
        return d2.min(1).mean()

    assert qerr(cent) < 0.7 * qerr(rand_cent)


def test_ivfpq_pipeline_golden(golden, small):
    base, q = small
    ix = OracleIVFPQ(32, 32, 8)
    ix.train(base[:1500], seed=42)
    assert np.array_equal(ix.centroids, golden["ivfpq_centroids"])
    assert np.array_equal(ix.codebooks, golden["ivfpq_codebooks"])
    ix.add(base)
    pd_, pl = ix.coarse_assign(q, 8)
    assert np.array_equal(pl, golden["ivfpq_probe_l"])
    sd, si = ix.search(q, 5, nprobe=8)
    assert np.array_equal(si, golden["ivfpq_i"])
    assert np.array_equal(sd, golden["ivfpq_d"])


def test_ivfpq_reference_recall_floors():
    """Reference gates (test_vector_index_ivfpq.py:106-111): with
    nprobe>10: recall@k >= 0.9; recall@1 >= 0.6; recall@10 >= 0.9 —
    here with the exact-rerank leg of the path (ivfpq.cc:675-726),
    which the reference clients enable via recall_num."""
    base = gen_clustered(20000, 64, seed=42, ncl=200)
    q = gen_queries(base, 64, seed=1)
    _, gti = flat_topk_f64(base, q, 100)
    ix = OracleIVFPQ(64, 64, 16)
    ix.train(base[:8000])
    ix.add(base)
    rd, ri = ix.search(q, 100, nprobe=16)
    # exact rerank of recall_num=100 candidates
    out = np.full((64, 10), -1, dtype=np.int64)
    r1 = np.full((64, 1), -1, dtype=np.int64)
    for t in range(q.shape[0]):
        cand = ri[t][ri[t] >= 0]
        ex = ((q[t].astype(np.float64)
               - base[cand].astype(np.float64)) ** 2).sum(1)
        srt = cand[np.lexsort((cand, ex))]
        out[t, :min(10, len(srt))] = srt[:10]
        r1[t, 0] = srt[0]
    assert recall_at(gti, r1, 1) >= 0.6
    assert recall_at(gti, out, 10) >= 0.9


def test_ivfpq_delete_never_returned(small):
    base, q = small
    ix = OracleIVFPQ(32, 32, 8)
    ix.train(base[:1500])
    ix.add(base)
    # mark every 3rd id deleted via bit 63 (realtime_mem_data.h:26)
    ids = ix.ids.copy()
    mask = (ids % 3 == 0)
    ix.ids = np.where(mask, ids | np.int64(-2 ** 63), ids)
    _, si = ix.search(q, 10, nprobe=32)
    live = si[si >= 0]
    assert (live % 3 != 0).all()
    # same via the IsValid bitmap with undamaged ids
    ix.ids = ids
    bm = np.zeros((2000 + 7) // 8, dtype=np.uint8)
    for vid in range(0, 2000, 3):
        bm[vid >> 3] |= 1 << (vid & 7)
    _, si2 = ix.search(q, 10, nprobe=32, del_bitmap=bm)
    live2 = si2[si2 >= 0]
    assert (live2 % 3 != 0).all()


def test_topk_tie_break_by_id():
    """Duplicated vectors -> exact distance ties -> ids ascending
    (the (dist,id) total order this rebuild defines)."""
    d = 16
    v = np.ones((1, d), dtype=np.float32) * 0.5
    base = np.repeat(v, 20, axis=0)
    q = v.copy()
    fd, fi = flat_search(base, q, 8, "L2")
    assert np.array_equal(fi[0], np.arange(8))
    assert np.allclose(fd[0], 0.0)


def test_empty_and_ragged_inputs():
    base = gen_clustered(100, 16, seed=3, ncl=5)
    q = gen_queries(base, 4, seed=4)
    # k > n: pad with -1
    fd, fi = flat_search(base[:3], q, 8, "L2")
    assert (fi[:, 3:] == -1).all() and (fd[:, 3:] == -1).all()
    # empty lists: search with probes pointing at empty lists
    ix = OracleIVFPQ(16, 8, 4)
    ix.train(base, seed=0)
    ix.add(base[:0] if False else base)  # all data
    probes = np.full((4, 3), -1, dtype=np.int64)  # key<0 skipped (cc:639)
    sd, si = ix.search(q, 5, nprobe=3, probes=probes)
    assert (si == -1).all()


def test_pct1_tables_match_numpy():
    """Decomposed tables (use_precomputed_table=1, ivfpq.h:254-262):
    A + B must equal the direct residual table to fp32 rounding, and the
    pct1 search must agree with the direct-mode search on recall."""
    import ctypes
    from oracle.gamma_oracle import _fp, _c
    d, M, ksub, dsub = 32, 8, 256, 4
    rng = np.random.default_rng(5)
    q = rng.random(d, dtype=np.float32)
    cent = rng.random(d, dtype=np.float32)
    books = rng.standard_normal((M, ksub, dsub)).astype(np.float32) * 0.2
    lib = RefLib.lib()
    A = np.empty((M, ksub), dtype=np.float32)
    B = np.empty((M, ksub), dtype=np.float32)
    lib.oracle_pct1_a_table.argtypes = lib.oracle_adc_table_ip.argtypes
    lib.oracle_pct1_b_table.argtypes = lib.oracle_adc_table_ip.argtypes
    lib.oracle_pct1_a_table(d, M, ksub, _fp(_c(q, np.float32)),
                            _fp(_c(books, np.float32)), _fp(A))
    lib.oracle_pct1_b_table(d, M, ksub, _fp(_c(cent, np.float32)),
                            _fp(_c(books, np.float32)), _fp(B))
    # numpy f64 check: dis0 + A + B ~ ||(q-c)_m - cw||^2
    qf, cf, bf = (x.astype(np.float64) for x in (q, cent, books))
    lib.oracle_l2_gemm_form.restype = ctypes.c_float
    dis0 = lib.oracle_l2_gemm_form(_fp(_c(q, np.float32)),
                                   _fp(_c(cent, np.float32)), d)
    for m in (0, 3, 7):
        for j in (0, 100, 255):
            want = (((qf - cf)[m * dsub:(m + 1) * dsub] - bf[m, j]) ** 2
                    ).sum() + ((qf - cf) ** 2).sum() \
                - np.dot(qf - cf, qf - cf)
            got = dis0 + A[m, j] + B[m, j] - ((qf - cf) ** 2).sum()
            # compare the per-subquantizer term alone
            term = dis0 + A[m, j] + B[m, j]
            direct = (((qf - cf)[m * dsub:(m + 1) * dsub] - bf[m, j]) ** 2
                      ).sum() + ((qf - cf) ** 2).sum() \
                - (((qf - cf)[m * dsub:(m + 1) * dsub]) ** 2).sum()
            assert abs(term - direct) < 1e-4, (m, j, term, direct)


def test_pct1_search_recall_equivalent():
    base = gen_clustered(8000, 32, seed=11, ncl=60)
    q = gen_queries(base, 32, seed=12)
    ix = OracleIVFPQ(32, 32, 8)
    ix.train(base[:4000])
    ix.add(base)
    d0, i0 = ix.search(q, 10, nprobe=8)
    d1, i1 = ix.search_pct1(q, 10, 8)
    # same probes; fp32 rounding differs -> allow small rank churn
    overlap = np.mean([len(set(a) & set(b)) / 10
                       for a, b in zip(i0.tolist(), i1.tolist())])
    assert overlap >= 0.9, overlap
    _, gti = flat_topk_f64(base, q, 10)
    assert abs(recall_at(gti, i0, 10) - recall_at(gti, i1, 10)) < 0.05


def test_pq_encode_tie_lowest_index():
    """Equal-distance codewords: the encoder must pick the LOWEST index
    (faiss/our argmin scans first-minimum). Pinned with duplicated
    codewords so ties are guaranteed."""
    d, M, ksub = 8, 2, 4
    rng = np.random.default_rng(5)
    books = rng.standard_normal((M, ksub, d // M)).astype(np.float32)
    books[0][2] = books[0][0]  # duplicate codeword: index 0 must win
    books[1][3] = books[1][1]  # index 1 must win over 3
    x = (books[0][0].tolist() + books[1][1].tolist())
    x = np.array([x], dtype=np.float32)
    codes = np.empty((1, M), dtype=np.uint8)
    RefLib.lib().oracle_pq_encode(
        1, d, M, ksub, _fp(_c(x, np.float32)),
        _fp(_c(books, np.float32)), _up8(codes))
    assert codes[0, 0] == 0
    assert codes[0, 1] == 1


def test_kmeans_deterministic():
    """Same data + seed -> identical centroids (training is part of the
    dump/load parity surface)."""
    x = gen_clustered(3000, 16, seed=3, ncl=20)
    c1 = kmeans(x, 8, niter=5, seed=9)
    c2 = kmeans(x, 8, niter=5, seed=9)
    assert np.array_equal(c1, c2)
    c3 = kmeans(x, 8, niter=5, seed=10)
    assert not np.array_equal(c1, c3)


def test_flat_search_rejects_unknown_metric():
    base = gen_clustered(100, 8, seed=1, ncl=4)
    q = gen_queries(base, 2, seed=2)
    with pytest.raises(ValueError):
        flat_search(base, q, 5, "cosine")
