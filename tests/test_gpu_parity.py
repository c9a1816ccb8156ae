"""GPU parity tests (run on a real MI355X via gpurun): pin each stage of
the HIP engine against the CPU oracle (oracle/ref_scan.c canonical
arithmetic), restating the reference's own gates:
  - FLAT: bit-exact (ids AND fp32 distances) — test_vector_index_flat.py
    gate is exactness.
  - IVFPQ: bit-exact vs oracle given the same trained model + probes;
    recall floors from test_vector_index_ivfpq.py:106-111.
  - deletes/updates, kill switch, dump/load.
"""
import ctypes
import os

import numpy as np
import pytest

import oracle as orc
from oracle.gamma_oracle import RefLib, _c, _fp

pytestmark = pytest.mark.gpu


def make_engine(tmp, **kw):
    from vearch_amd import GammaEngine
    return GammaEngine(path=str(tmp), **kw)


@pytest.fixture(scope="module")
def data():
    base = orc.gen_clustered(20000, 64, seed=42, ncl=200)
    q = orc.gen_queries(base, 64, seed=1)
    return base, q


# ---------------------------------------------------------------- FLAT
def test_flat_bitexact_small(tmp_path, data):
    base, q = data
    eng = make_engine(tmp_path)
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}')
    eng.add(base)
    gd, gi = eng.raw_search(q, 10)
    od, oi = orc.flat_search(base, q, 10, "L2")
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    eng.close()


def test_flat_bitexact_ip(tmp_path, data):
    base, q = data
    eng = make_engine(tmp_path)
    eng.create_table(64, "FLAT", '{"metric_type": "InnerProduct"}')
    eng.add(base)
    gd, gi = eng.raw_search(q, 10)
    od, oi = orc.flat_search(base, q, 10, "InnerProduct")
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    eng.close()


def test_flat_gemm_path_bitexact(tmp_path):
    """nq >= 512 and N >= 200k takes the chunked MFMA GEMM + select path;
    results must still be bit-exact after the canonical re-rank."""
    base = orc.gen_clustered(220000, 32, seed=5, ncl=500)
    q = orc.gen_queries(base, 600, seed=6)
    eng = make_engine("/tmp/gamma_flat_gemm")
    eng.create_table(32, "FLAT", '{"metric_type": "L2"}')
    eng.add(base)
    gd, gi = eng.raw_search(q, 10)
    od, oi = orc.flat_search(base, q, 10, "L2")
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    eng.close()


def test_flat_deletes(tmp_path, data):
    base, q = data
    eng = make_engine(tmp_path)
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}')
    eng.add(base)
    for vid in range(0, 20000, 7):
        eng.delete_doc(str(vid))
    bm = np.zeros((20000 + 7) // 8, dtype=np.uint8)
    for vid in range(0, 20000, 7):
        bm[vid >> 3] |= 1 << (vid & 7)
    gd, gi = eng.raw_search(q, 10)
    od, oi = orc.flat_search(base, q, 10, "L2", del_bitmap=bm)
    assert np.array_equal(gi, oi)
    assert not any(v % 7 == 0 for v in gi.ravel().tolist() if v >= 0)
    eng.close()


# ---------------------------------------------------------------- IVFPQ
def _oracle_from_engine(eng, d, nlist, M, metric="L2"):
    ox = orc.OracleIVFPQ(d, nlist, M, metric=metric)
    cent, books = eng.debug_model(nlist, d, M)
    ox.centroids, ox.codebooks = cent, books
    ids_all, codes_all, offsets = [], [], [0]
    for ln in range(nlist):
        li, lc = eng.debug_list(ln, M)
        ids_all.append(li)
        codes_all.append(lc)
        offsets.append(offsets[-1] + len(li))
    ox.ids = np.concatenate(ids_all)
    ox.codes = np.concatenate(codes_all)
    ox.offsets = np.array(offsets, dtype=np.int64)
    return ox


@pytest.fixture(scope="module")
def ivfpq_engine(data):
    base, q = data
    eng = make_engine("/tmp/gamma_ivfpq_mod")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    yield eng
    eng.close()


def test_ivfpq_bucket_population(ivfpq_engine):
    total = 0
    for ln in range(64):
        li, _ = ivfpq_engine.debug_list(ln, 16)
        total += len(li)
    assert total == 20000


def test_ivfpq_adc_bitexact_vs_oracle(data, ivfpq_engine):
    base, q = data
    eng = ivfpq_engine
    gd, gi = eng.raw_search(q, 10, nprobe=16)
    ox = _oracle_from_engine(eng, 64, 64, 16)
    pdists, probes = eng.debug_coarse_assign(q, 16)
    od, oi = ox.search_pct1(q, 10, 16, probes=probes, probe_dists=pdists)
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)


def test_ivfpq_coarse_assign_close_to_oracle(data, ivfpq_engine):
    """GEMM-form distances round differently; require same probe sets
    wherever the oracle's distance gap exceeds fp32 noise."""
    base, q = data
    eng = ivfpq_engine
    gdist, glists = eng.debug_coarse_assign(q, 8)
    ox = _oracle_from_engine(eng, 64, 64, 16)
    odist, olists = ox.coarse_assign(q, 8)
    assert np.allclose(gdist, odist, rtol=1e-4, atol=1e-4)
    agree = (glists == olists).mean()
    assert agree > 0.95, f"probe agreement too low: {agree}"


def test_ivfpq_rerank_exact_distances(data, ivfpq_engine):
    base, q = data
    eng = ivfpq_engine
    gd, gi = eng.raw_search(q, 10, nprobe=16, rerank=100)
    lib = RefLib.lib()
    lib.oracle_l2sqr.restype = ctypes.c_float
    for t in range(q.shape[0]):
        for j in range(10):
            if gi[t, j] < 0:
                continue
            want = lib.oracle_l2sqr(_fp(_c(q[t], np.float32)),
                                    _fp(_c(base[gi[t, j]], np.float32)), 64)
            assert gd[t, j] == want, (t, j)
        # sorted ascending, ties by id
        row = [(gd[t, j], gi[t, j]) for j in range(10) if gi[t, j] >= 0]
        assert row == sorted(row)


def test_ivfpq_recall_floors(data, ivfpq_engine):
    """Reference gates (test_vector_index_ivfpq.py:106-111) with the
    rerank leg on: R@1>=0.6, R@10>=0.9."""
    base, q = data
    eng = ivfpq_engine
    _, gti = orc.flat_topk_f64(base, q, 100)
    gd, gi = eng.raw_search(q, 10, nprobe=16, rerank=100)
    r1 = orc.recall_at(gti, gi[:, :1], 1)
    r10 = orc.recall_at(gti, gi, 10)
    assert r1 >= 0.6, f"recall@1={r1}"
    assert r10 >= 0.9, f"recall@10={r10}"


def test_ivfpq_delete_and_update(data):
    """Own engine: mutates state (the shared ivfpq_engine stays clean so
    later pb tests can equate _id strings with docids)."""
    base, q = data
    eng = make_engine("/tmp/gamma_ivfpq_mut")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    gd0, gi0 = eng.raw_search(q[:4], 5, nprobe=64)
    victim = int(gi0[0, 0])
    eng.delete_doc(str(victim))
    gd1, gi1 = eng.raw_search(q[:4], 5, nprobe=64)
    assert victim not in gi1[0].tolist()
    # re-add via the Doc path: becomes a new docid, old stays dead
    eng.add_doc(str(victim), base[victim])
    gd2, gi2 = eng.raw_search(q[:4], 5, nprobe=64)
    new_id = eng.num_docs() - 1
    assert new_id in gi2[0].tolist()
    eng.close()


# ---------------------------------------------------------------- IVFFLAT
def test_ivfflat_bitexact_vs_oracle(data):
    base, q = data
    eng = make_engine("/tmp/gamma_ivfflat")
    eng.create_table(
        64, "IVFFLAT",
        '{"ncentroids": 64, "metric_type": "L2", '
        '"training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    gd, gi = eng.raw_search(q, 10, nprobe=16)
    # oracle with the engine's lists + probes
    cent, _ = eng.debug_model(64, 64, 0)
    ids_all, vec_all, offsets = [], [], [0]
    for ln in range(64):
        li, lc = eng.debug_list(ln, 64 * 4)
        ids_all.append(li)
        vec_all.append(lc.view(np.float32).reshape(-1, 64))
        offsets.append(offsets[-1] + len(li))
    ids = np.concatenate(ids_all)
    vecs = np.ascontiguousarray(np.concatenate(vec_all), dtype=np.float32)
    offsets = np.array(offsets, dtype=np.int64)
    _, probes = eng.debug_coarse_assign(q, 16)
    lib = RefLib.lib()
    from oracle.gamma_oracle import _ip64, _up8
    od = np.empty((64, 10), dtype=np.float32)
    oi = np.empty((64, 10), dtype=np.int64)
    lib.oracle_ivfflat_search(
        64, 64, 64, _fp(_c(q, np.float32)), _ip64(offsets), _ip64(ids),
        _fp(vecs), 16, _ip64(_c(probes, np.int64)), _up8(None), 0, 10,
        _fp(od), _ip64(oi))
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    eng.close()


# ----------------------------------------------------------- C-ABI Search
def test_search_pb_end_to_end(data, ivfpq_engine):
    base, q = data
    eng = ivfpq_engine
    res = eng.search_pb(q[:8], topn=5,
                        index_params='{"nprobe": 16, "recall_num": 50}')
    gd, gi = eng.raw_search(q[:8], 5, nprobe=16, rerank=50)
    assert len(res) == 8
    for t in range(8):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        scores = [it["score"] for it in res[t]["items"]]
        want = [i for i in gi[t].tolist() if i >= 0]
        assert ids == want
        assert np.allclose(scores, gd[t][:len(scores)])


def test_search_pb_l2_sqrt(data, ivfpq_engine):
    base, q = data
    eng = ivfpq_engine
    res = eng.search_pb(q[:4], topn=3, index_params='{"nprobe": 16}',
                        l2_sqrt=True)
    gd, gi = eng.raw_search(q[:4], 3, nprobe=16)
    for t in range(4):
        scores = [it["score"] for it in res[t]["items"]]
        assert np.allclose(scores, np.sqrt(gd[t][:len(scores)]), rtol=1e-6)


def test_kill_switch(data, ivfpq_engine):
    from vearch_amd import clear_kill, set_kill
    base, q = data
    eng = ivfpq_engine
    set_kill("killme", 1)
    with pytest.raises((InterruptedError, RuntimeError)):
        eng.search_pb(q[:4], topn=3, request_id="killme", partition_id=1)
    clear_kill("killme", 1)
    res = eng.search_pb(q[:4], topn=3, request_id="killme", partition_id=1)
    assert len(res) == 4


def test_query_and_get_doc(data, ivfpq_engine):
    eng = ivfpq_engine
    import ctypes as c
    from vearch_amd.engine import lib
    out = c.c_char_p()
    n = c.c_int()
    rc = lib().GetDocByID(eng.h, b"123", 3, c.byref(out), c.byref(n))
    assert rc == 0
    buf = c.string_at(out, n.value)
    from vearch_amd.fbsenc import DATA_VECTOR
    # parse with our own fbs reader via the C++ roundtrip hook
    L = lib()
    L.GammaTestDocRoundtrip.argtypes = [c.c_char_p, c.c_int,
                                        c.POINTER(c.c_char_p),
                                        c.POINTER(c.c_int)]
    out2 = c.c_char_p()
    n2 = c.c_int()
    assert L.GammaTestDocRoundtrip(buf, len(buf), c.byref(out2),
                                   c.byref(n2)) == 0


def test_dump_load_roundtrip(data):
    base, q = data
    path = "/tmp/gamma_dumpload"
    os.makedirs(path, exist_ok=True)
    eng = make_engine(path)
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 4000}')
    eng.add(base[:8000])
    eng.build_index()
    gd0, gi0 = eng.raw_search(q, 10, nprobe=32)
    eng.dump()
    eng.close()
    eng2 = make_engine(path)
    eng2.create_table(
        64, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 4000}')
    eng2.load()
    assert eng2.num_docs() == 8000
    gd1, gi1 = eng2.raw_search(q, 10, nprobe=32)
    assert np.array_equal(gi0, gi1)
    assert np.array_equal(gd0, gd1)
    eng2.close()


def test_untrained_falls_back_to_flat(data):
    base, q = data
    eng = make_engine("/tmp/gamma_untrained")
    eng.create_table(64, "IVFPQ",
                     '{"ncentroids": 64, "nsubvector": 16, '
                     '"metric_type": "L2"}')
    eng.add(base[:3000])
    # no build_index: Search must brute-force (engine.cc:286-288 analog)
    gd, gi = eng.raw_search(q, 10)
    od, oi = orc.flat_search(base[:3000], q, 10, "L2")
    assert np.array_equal(gi, oi)
    eng.close()


def test_ivfpq_north_star_shape_recall():
    """d=128 m=32 (the north-star shape) at small N: recall gate >=0.9
    with rerank; BASELINE's >=0.95 is checked in bench at full size."""
    base = orc.gen_clustered(50000, 128, seed=42, ncl=500)
    q = orc.gen_queries(base, 64, seed=2)
    _, gti = orc.flat_topk_f64(base, q, 100)
    eng = make_engine("/tmp/gamma_ns")
    eng.create_table(
        128, "IVFPQ",
        '{"ncentroids": 256, "nsubvector": 32, "metric_type": "L2", '
        '"training_threshold": 20000}')
    eng.add(base)
    eng.build_index()
    _, gi = eng.raw_search(q, 10, nprobe=32, rerank=200)
    r10 = orc.recall_at(gti, gi, 10)
    assert r10 >= 0.9, f"recall@10={r10}"
    eng.close()


def test_ivfpq_ip_bitexact_vs_oracle(data):
    """InnerProduct IVFPQ: spherical k-means + query-level IP table +
    dis0 = q.c (gamma_index_ivfpq.h:164-167, 223-236) — bit-exact vs the
    oracle on the same model/probes."""
    base, q = data
    eng = make_engine("/tmp/gamma_ivfpq_ip")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, '
        '"metric_type": "InnerProduct", "training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    gd, gi = eng.raw_search(q, 10, nprobe=16)
    ox = _oracle_from_engine(eng, 64, 64, 16, metric="InnerProduct")
    _, probes = eng.debug_coarse_assign(q, 16)
    od, oi = ox.search(q, 10, 16, probes=probes)
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    # IP similarity is descending
    for t in range(q.shape[0]):
        row = [gd[t, j] for j in range(10) if gi[t, j] >= 0]
        assert row == sorted(row, reverse=True)
    eng.close()


def test_edge_cases(data):
    base, q = data
    eng = make_engine("/tmp/gamma_edge")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}')
    # search an empty engine: all -1
    gd, gi = eng.raw_search(q[:4], 5)
    assert (gi == -1).all()
    # k > n
    eng.add(base[:3])
    gd, gi = eng.raw_search(q[:4], 8)
    assert (gi[:, 3:] == -1).all()
    assert set(gi[0, :3].tolist()) == {0, 1, 2}
    eng.close()
    # nprobe > nlist clamps (ivfpq.cc:572-580 semantics)
    eng2 = make_engine("/tmp/gamma_edge2")
    eng2.create_table(
        64, "IVFPQ",
        '{"ncentroids": 16, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 2000}')
    eng2.add(base[:4000])
    eng2.build_index()
    gd, gi = eng2.raw_search(q[:4], 5, nprobe=999)
    assert (gi[:, :5] >= 0).all()
    # unknown metric_type is a parse error, not a silent InnerProduct
    with pytest.raises(RuntimeError, match="metric_type"):
        eng2.search_pb(q[:2], topn=5,
                       index_params='{"metric_type": "cosine"}')
    # k2 over the selector cap names the limit instead of "search failed"
    with pytest.raises(RuntimeError, match="1024"):
        eng2.search_pb(q[:2], topn=1100)
    with pytest.raises(RuntimeError, match="1024"):
        eng2.search_pb(q[:2], topn=5,
                       index_params='{"recall_num": 2000}')
    eng2.close()


def test_delete_all_then_search(data):
    base, q = data
    eng = make_engine("/tmp/gamma_delall")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}')
    eng.add(base[:100])
    for vid in range(100):
        eng.delete_doc(str(vid))
    gd, gi = eng.raw_search(q[:4], 5)
    assert (gi == -1).all()
    eng.close()


def test_filtered_search_term_and_range(data):
    """scalar filters -> device exclusion bitmap. Term and
    range filters through the real protobuf Search; parity vs the oracle
    FLAT scan with an equivalent bitmap."""
    import struct
    from vearch_amd import fbsenc
    base, q = data
    eng = make_engine("/tmp/gamma_filter")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     scalar_fields=[("tag", fbsenc.DATA_STRING),
                                    ("num", fbsenc.DATA_INT)])
    n = 3000
    for vid in range(n):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", b"grp%d" % (vid % 4),
                             fbsenc.DATA_STRING),
                            ("num", struct.pack("<i", vid % 100),
                             fbsenc.DATA_INT)])
    # term filter tag == grp1
    res = eng.search_pb(q[:8], topn=10,
                        term_filters=[("tag", b"grp1")])
    bm = np.zeros((n + 7) // 8, dtype=np.uint8)
    for vid in range(n):
        if vid % 4 != 1:
            bm[vid >> 3] |= 1 << (vid & 7)
    od, oi = orc.flat_search(base[:n], q[:8], 10, "L2", del_bitmap=bm)
    for t in range(8):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert ids == [i for i in oi[t].tolist() if i >= 0]
        assert all(i % 4 == 1 for i in ids)
    # term filter with two \x01-separated values (grp1 or grp2)
    res = eng.search_pb(q[:4], topn=10,
                        term_filters=[("tag", b"grp1\x01grp2")])
    for t in range(4):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert all(i % 4 in (1, 2) for i in ids)
    # range filter 10 <= num <= 20
    res = eng.search_pb(
        q[:8], topn=10,
        range_filters=[("num", struct.pack("<i", 10),
                        struct.pack("<i", 20), True, True)])
    bm2 = np.zeros((n + 7) // 8, dtype=np.uint8)
    for vid in range(n):
        if not (10 <= vid % 100 <= 20):
            bm2[vid >> 3] |= 1 << (vid & 7)
    od2, oi2 = orc.flat_search(base[:n], q[:8], 10, "L2", del_bitmap=bm2)
    for t in range(8):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert ids == [i for i in oi2[t].tolist() if i >= 0]
    # combined AND + unknown field error
    res = eng.search_pb(
        q[:4], topn=10, term_filters=[("tag", b"grp1")],
        range_filters=[("num", struct.pack("<i", 10),
                        struct.pack("<i", 40), True, False)])
    for t in range(4):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert all(i % 4 == 1 and 10 <= i % 100 < 40 for i in ids)
    with pytest.raises(RuntimeError):
        eng.search_pb(q[:2], topn=5, term_filters=[("nope", b"x")])
    eng.close()


def test_field_index_lifecycle_and_backup(data):
    """AddFieldIndexWithParams / RemoveFieldIndex route SCALAR names
    onto the ScalarFieldIndex machinery (engine.cc:1561,1648 semantics:
    duplicate add and unknown remove are OK no-ops, unknown fields are
    errors); Backup(0) creates a full dump under <path>/backup
    (engine.cc:1529)."""
    import struct
    from vearch_amd import fbsenc
    base, q = data
    path = "/tmp/gamma_fieldindex"
    eng = make_engine(path)
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     scalar_fields=[("tag", fbsenc.DATA_STRING),
                                    ("num", fbsenc.DATA_INT)])
    for vid in range(500):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", b"grp%d" % (vid % 3),
                             fbsenc.DATA_STRING),
                            ("num", struct.pack("<i", vid),
                             fbsenc.DATA_INT)])
    # scalar index on both fields; duplicate add ignored; vector ok
    eng.add_field_index("ix_tag", ["tag"], "SCALAR")
    eng.add_field_index("ix_tag", ["tag"], "SCALAR")  # no-op
    eng.add_field_index("ix_num", ["num"], "SCALAR")
    eng.add_field_index("ix_vec", [eng.vec_name], "IVFPQ")
    with pytest.raises(RuntimeError, match="not found"):
        eng.add_field_index("ix_bad", ["nope"], "SCALAR")
    # filters work through the (now eagerly built) indexes
    res = eng.search_pb(q[:4], topn=10, term_filters=[("tag", b"grp1")])
    for t in range(4):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert ids and all(i % 3 == 1 for i in ids)
    # remove + idempotent remove; filters still correct (lazy rebuild)
    eng.remove_field_index("ix_tag")
    eng.remove_field_index("ix_tag")
    eng.remove_field_index("never_existed")
    res = eng.search_pb(q[:2], topn=10, term_filters=[("tag", b"grp2")])
    for t in range(2):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert ids and all(i % 3 == 2 for i in ids)
    # backup: command 0 writes a loadable dump under <path>/backup
    import os
    eng.backup(0)
    assert os.path.exists(os.path.join(path, "backup", "gamma.dump"))
    eng.backup(1)  # non-create commands: accepted no-ops
    eng.close()


def test_filter_not_and_or(data):
    """FilterOperator semantics (scalar_index_types.h:44 And=0 Or=1
    Not=2): per-filter Not on terms (BitmapIndex::NotIn — docs missing
    the value, including docs with no match at all, pass), numeric
    NotEqual (Not + equal inclusive bounds, bitmap_index.cc:196), and
    the request-level `operator` OR union
    (scalar_index_manager.cc:1188-1190)."""
    import struct
    from vearch_amd import fbsenc
    base, q = data
    eng = make_engine("/tmp/gamma_filter_notor")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     scalar_fields=[("tag", fbsenc.DATA_STRING),
                                    ("num", fbsenc.DATA_INT)])
    n = 2000
    for vid in range(n):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", b"grp%d" % (vid % 4),
                             fbsenc.DATA_STRING),
                            ("num", struct.pack("<i", vid % 50),
                             fbsenc.DATA_INT)])

    def check(res, nq, pred):
        for t in range(nq):
            keep = [vid for vid in range(n) if pred(vid)]
            bm = np.zeros((n + 7) // 8, dtype=np.uint8)
            for vid in range(n):
                if not pred(vid):
                    bm[vid >> 3] |= 1 << (vid & 7)
            od, oi = orc.flat_search(base[:n], q[t:t + 1], 10, "L2",
                                     del_bitmap=bm)
            ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
            assert ids == [i for i in oi[0].tolist() if i >= 0]
            assert all(pred(i) for i in ids)

    # NOT-IN term filter (is_union=2): everything except grp1/grp2
    res = eng.search_pb(q[:4], topn=10,
                        term_filters=[("tag", b"grp1\x01grp2", 2)])
    check(res, 4, lambda v: v % 4 not in (1, 2))
    # numeric NotEqual: num != 7
    seven = struct.pack("<i", 7)
    res = eng.search_pb(q[:4], topn=10,
                        range_filters=[("num", seven, seven, True, True,
                                        2)])
    check(res, 4, lambda v: v % 50 != 7)
    # request-level OR: tag==grp1 OR 10<=num<=12
    res = eng.search_pb(
        q[:4], topn=10, operator=1,
        term_filters=[("tag", b"grp1")],
        range_filters=[("num", struct.pack("<i", 10),
                        struct.pack("<i", 12), True, True)])
    check(res, 4, lambda v: v % 4 == 1 or 10 <= v % 50 <= 12)
    # AND of a Not with a plain term: tag==grp1 AND num != 7
    res = eng.search_pb(
        q[:4], topn=10, term_filters=[("tag", b"grp1")],
        range_filters=[("num", seven, seven, True, True, 2)])
    check(res, 4, lambda v: v % 4 == 1 and v % 50 != 7)
    # Not filter through the Query browse path too
    docs = eng.query_pb(term_filters=[("tag", b"grp0\x01grp1\x01grp2", 2)],
                        limit=50)
    ids = [int(it["fields"]["_id"]) for it in docs[0]["items"]]
    assert ids == [v for v in range(n) if v % 4 == 3][:50]
    eng.close()


def test_filter_index_incremental_and_stringarray(data):
    """The scalar indexes append lazily: filter, add more docs (+ one
    update and one delete), filter again — results must track the
    mutations exactly. STRINGARRAY any-element matching included."""
    import struct
    from vearch_amd import fbsenc
    base, q = data
    eng = make_engine("/tmp/gamma_filter_inc")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     scalar_fields=[("tags", fbsenc.DATA_STRINGARRAY),
                                    ("num", fbsenc.DATA_INT)])

    def addd(vid, tags, num):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tags", tags, fbsenc.DATA_STRINGARRAY),
                            ("num", struct.pack("<i", num),
                             fbsenc.DATA_INT)])

    for vid in range(1000):
        addd(vid, b"a\x01b" if vid % 2 == 0 else b"c", vid)
    res = eng.search_pb(q[:4], topn=10, term_filters=[("tags", b"b")])
    for t in range(4):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert ids and all(i % 2 == 0 for i in ids)
    # incremental rows after the first (index-building) filter call
    for vid in range(1000, 2000):
        addd(vid, b"b", vid)
    addd(7, b"zzz", 7)  # update: pkey "7" re-keyed, old row dead
    eng.delete_doc("1002")
    res = eng.search_pb(q[:4], topn=300, term_filters=[("tags", b"b")])
    got = set()
    for t in range(4):
        got |= {int(it["fields"]["_id"]) for it in res[t]["items"]}
    assert all((v < 1000 and v % 2 == 0) or v >= 1000 for v in got)
    assert 1002 not in got          # deleted
    assert 7 not in got             # updated away from tag b
    assert any(v >= 1000 for v in got)  # new rows visible
    # numeric range over the grown column
    res = eng.search_pb(
        q[:4], topn=300,
        range_filters=[("num", struct.pack("<i", 990),
                        struct.pack("<i", 1010), True, True)])
    for t in range(4):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert ids and all(990 <= i <= 1010 and i != 1002 for i in ids)
    eng.close()


def test_filtered_search_ivfpq(data, ivfpq_engine):
    """filters work on the IVFPQ path too (same bitmap arg)."""
    base, q = data
    eng = make_engine("/tmp/gamma_filter_pq")
    from vearch_amd import fbsenc
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000}',
        scalar_fields=[("tag", fbsenc.DATA_STRING)])
    import struct
    for vid in range(20000):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", b"odd" if vid % 2 else b"even",
                             fbsenc.DATA_STRING)])
    eng.build_index()
    res = eng.search_pb(q[:8], topn=10,
                        index_params='{"nprobe": 64, "recall_num": 100}',
                        term_filters=[("tag", b"odd")])
    for t in range(8):
        ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert len(ids) > 0
        assert all(i % 2 == 1 for i in ids)
    eng.close()


def test_query_pb_by_ids_and_filters(data):
    from vearch_amd import fbsenc
    base, q = data
    eng = make_engine("/tmp/gamma_query")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     scalar_fields=[("tag", fbsenc.DATA_STRING)])
    for vid in range(200):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", b"a" if vid % 2 else b"b",
                             fbsenc.DATA_STRING)])
    res = eng.query_pb(document_ids=["5", "7", "nope"], fields=("_id", "tag"))
    ids = [int(it["fields"]["_id"]) for it in res[0]["items"]]
    assert ids == [5, 7]
    assert res[0]["items"][0]["fields"]["tag"] == b"a"
    # filtered browse with limit (Engine::Query filter path)
    res = eng.query_pb(term_filters=[("tag", b"b")], limit=10,
                       fields=("_id", "tag"))
    ids = [int(it["fields"]["_id"]) for it in res[0]["items"]]
    assert len(ids) == 10
    assert all(i % 2 == 0 for i in ids)
    eng.close()


def test_concurrent_search_and_mutation(data):
    """The reference allows Search from arbitrary cgo threads while a
    background thread mutates (engine.cc:1108-1127). Searches run
    concurrently (read lock + per-search scratch/stream pool); mutators
    take the write lock — results must stay sane under contention."""
    import threading
    base, q = data
    eng = make_engine("/tmp/gamma_conc")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    errors = []

    def searcher(tid):
        try:
            for _ in range(20):
                gd, gi = eng.raw_search(q[tid:tid + 8], 5, nprobe=16)
                live = gi[gi >= 0]
                assert (live < 30000).all()
        except Exception as ex:  # noqa: BLE001
            errors.append(ex)

    threads = [threading.Thread(target=searcher, args=(t,))
               for t in range(4)]
    for t in threads:
        t.start()
    for vid in range(0, 2000, 3):
        eng.delete_doc(str(vid))
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors
    # after the dust settles: deleted ids gone
    gd, gi = eng.raw_search(q[:8], 10, nprobe=64)
    live = gi[gi >= 0]
    bad = [v for v in live.tolist() if v < 2000 and v % 3 == 0]
    assert not bad
    eng.close()


def test_scan_launch_knobs_bitexact(data, ivfpq_engine):
    """The perf-experiment knobs (GAMMA_ADC_C staging depth,
    GAMMA_SCAN_BS block size, GAMMA_SCAN_S probe split, GAMMA_QSORT
    query schedule) must not change results: the selector's top-k is
    exact under the (dist,id) total order and outputs are written at
    the original query row."""
    import os
    base, q = data
    eng = ivfpq_engine
    # >= 2048 queries so GAMMA_QSORT actually activates its schedule
    qq = np.repeat(q, 32, axis=0)
    ref_d, ref_i = eng.raw_search(qq, 10, nprobe=16, rerank=64)
    knobs = [
        {"GAMMA_ADC_C": "2"},
        {"GAMMA_SCAN_BS": "256"},
        {"GAMMA_SCAN_S": "2"},
        {"GAMMA_QSORT": "1"},
        {"GAMMA_ADC_C": "2", "GAMMA_SCAN_BS": "256", "GAMMA_SCAN_S": "2"},
    ]
    for kv in knobs:
        for k, v in kv.items():
            os.environ[k] = v
        try:
            gd, gi = eng.raw_search(qq, 10, nprobe=16, rerank=64)
        finally:
            for k in kv:
                del os.environ[k]
        assert np.array_equal(gi, ref_i), kv
        assert np.array_equal(gd, ref_d), kv


def test_multi_vector_search(data):
    """Multi-vector-field search (vector_manager.cc:851-1090): per-field
    top-n, docid-intersection merge, WeightedRanker combined scores,
    docid-order vs multi_vector_rank score-order — checked against a
    python replica of the reference merge over the oracle's exact
    per-field FLAT results."""
    base, q = data  # d=64
    n, nq, topn, d2 = 3000, 5, 20, 32
    rng = np.random.default_rng(97)
    base2 = rng.standard_normal((n, d2)).astype(np.float32)
    q2 = base2[:nq] + 0.05 * rng.standard_normal((nq, d2)).astype(
        np.float32)
    eng = make_engine("/tmp/gamma_multivec")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     extra_vecs=[("emb2", d2)])
    for vid in range(n):
        eng.add_doc(str(vid), base[vid],
                    extra_vecs=[("emb2", base2[vid])])

    # expected: per-field exact results from the oracle, merged exactly
    # as the reference does (intersection, weights, order)
    d1o, i1o = orc.flat_search(base[:n], q[:nq], topn, "L2")
    d2o, i2o = orc.flat_search(base2, q2, topn, "L2")

    def merge(t, weights, by_score):
        l1 = sorted((int(i), float(d)) for i, d in zip(i1o[t], d1o[t])
                    if i >= 0)
        m2 = {int(i): float(d) for i, d in zip(i2o[t], d2o[t]) if i >= 0}
        out = []
        for i, dd in l1:
            if i in m2:
                score = dd * weights[0] + m2[i] * weights[1]
                out.append((score, i))
        if by_score:
            out.sort(key=lambda x: x[0])
        return out

    # score-ordered (multi_vector_rank), default weights 1/2
    res = eng.search_pb(q[:nq], topn=topn,
                        extra_vec_queries=[("emb2", q2)],
                        multi_vector_rank=1)
    for t in range(nq):
        exp = merge(t, (0.5, 0.5), True)
        got = [(it["score"], int(it["fields"]["_id"]))
               for it in res[t]["items"]]
        assert [g[1] for g in got] == [e[1] for e in exp]
        for g, e in zip(got, exp):
            assert abs(g[0] - e[0]) < 1e-6 * max(1.0, abs(e[0]))

    # docid-ordered (no multi_vector_rank)
    res = eng.search_pb(q[:nq], topn=topn,
                        extra_vec_queries=[("emb2", q2)])
    for t in range(nq):
        exp = merge(t, (0.5, 0.5), False)
        got_ids = [int(it["fields"]["_id"]) for it in res[t]["items"]]
        assert got_ids == [e[1] for e in exp]
        assert got_ids == sorted(got_ids)

    # WeightedRanker weights
    res = eng.search_pb(
        q[:nq], topn=topn, extra_vec_queries=[("emb2", q2)],
        multi_vector_rank=1,
        ranker='{"type": "WeightedRanker", "params": [0.9, 0.1]}')
    for t in range(nq):
        exp = merge(t, (0.9, 0.1), True)
        got = [(it["score"], int(it["fields"]["_id"]))
               for it in res[t]["items"]]
        assert [g[1] for g in got] == [e[1] for e in exp]
    # bad ranker length is an error (common_query_data.h:291)
    with pytest.raises(RuntimeError, match="length"):
        eng.search_pb(q[:2], topn=5, extra_vec_queries=[("emb2", q2[:2])],
                      ranker='{"type": "WeightedRanker", "params": [1.0]}')
    # unknown field name is an error
    with pytest.raises(RuntimeError, match="unknown vector field"):
        eng.search_pb(q[:2], topn=5,
                      extra_vec_queries=[("nope", q2[:2])])

    # doc fetch returns EVERY vector field of a multi-vector table
    docs = eng.query_pb(document_ids=["7"], is_vector_value=True)
    fields = docs[0]["items"][0]["fields"]
    got1 = np.frombuffer(fields[eng.vec_name], dtype=np.float32)
    got2 = np.frombuffer(fields["emb2"], dtype=np.float32)
    assert np.array_equal(got1, base[7])
    assert np.array_equal(got2, base2[7])

    # dump/load keeps the extra field (v2 dump format)
    eng.dump()
    eng.close()
    eng2 = make_engine("/tmp/gamma_multivec")
    eng2.create_table(64, "FLAT", '{"metric_type": "L2"}',
                      extra_vecs=[("emb2", d2)])
    eng2.load()
    res = eng2.search_pb(q[:nq], topn=topn,
                         extra_vec_queries=[("emb2", q2)],
                         multi_vector_rank=1)
    for t in range(nq):
        exp = merge(t, (0.5, 0.5), True)
        assert [int(it["fields"]["_id"]) for it in res[t]["items"]] == \
            [e[1] for e in exp]
    eng2.close()


def test_multi_vector_ivfpq(data):
    """Multi-vector over trained IVFPQ indexes: both fields carry the
    same vectors, so the merged ids must echo the single-field search's
    ids (intersection of identical top-n sets) with score = the
    weighted sum of the two identical distances."""
    base, q = data
    n, nq, topn = 6000, 4, 10
    eng = make_engine("/tmp/gamma_multivec_pq")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 6000}',
        extra_vecs=[("emb2", 64)])
    for vid in range(n):
        eng.add_doc(str(vid), base[vid], extra_vecs=[("emb2", base[vid])])
    eng.build_index()
    single = eng.search_pb(q[:nq], topn=topn,
                           index_params='{"recall_num": 50}')
    multi = eng.search_pb(q[:nq], topn=topn,
                          index_params='{"recall_num": 50}',
                          extra_vec_queries=[("emb2", q[:nq])],
                          multi_vector_rank=1)
    for t in range(nq):
        sids = [int(it["fields"]["_id"]) for it in single[t]["items"]]
        mids = [int(it["fields"]["_id"]) for it in multi[t]["items"]]
        assert sids, "single-field search returned nothing (vacuous)"
        assert mids == sids
        for si, mi in zip(single[t]["items"], multi[t]["items"]):
            assert abs(mi["score"] - si["score"]) < 1e-5 * max(
                1.0, abs(si["score"]))
    eng.close()


def test_score_range_docwalk_status(data, ivfpq_engine):
    """Request-surface corners: min/max score ranges (single AND
    per-field multi-vector), GetDocByDocID(next) walking over deleted
    docs, and the EngineStatus / MemoryInfo JSON."""
    import json as _json
    import ctypes as c
    from vearch_amd.engine import lib
    base, q = data
    eng = ivfpq_engine
    # single-path score range: min/max bracket the true score window
    full = eng.search_pb(q[:4], topn=10,
                         index_params='{"recall_num": 50}')
    for t in range(4):
        scores = [it["score"] for it in full[t]["items"]]
        assert scores == sorted(scores)
        lo, hi = scores[2], scores[6]
        ranged = eng.search_pb(q[t:t + 1], topn=10,
                               index_params='{"recall_num": 50}',
                               min_score=lo, max_score=hi)
        rs = [it["score"] for it in ranged[0]["items"]]
        assert rs and all(lo <= s <= hi for s in rs)
    # pagination: offset=o returns ranks [o, o+topn) of the full order
    full10 = eng.search_pb(q[:2], topn=10,
                           index_params='{"recall_num": 50}')
    for t in range(2):
        paged = eng.search_pb(q[t:t + 1], topn=5, offset=3,
                              index_params='{"recall_num": 50}')
        pids = [int(it["fields"]["_id"]) for it in paged[0]["items"]]
        fids = [int(it["fields"]["_id"]) for it in full10[t]["items"]]
        assert pids == fids[3:8]
    # status / memory JSON are well-formed and consistent
    st = _json.loads(eng.status())
    assert st["doc_count"] > 0 and st["index_status"] == 2
    out = c.c_char_p()
    ln = c.c_int()
    lib().GetMemoryInfo.argtypes = [c.c_void_p, c.POINTER(c.c_char_p),
                                    c.POINTER(c.c_int)]
    lib().GetMemoryInfo(eng.h, c.byref(out), c.byref(ln))
    mi = _json.loads(c.string_at(out, ln.value).decode())
    assert mi["vector_mem_bytes"] > 0
    # SetConfig/GetConfig round-trip (engine.cc:2071-2114 fields)
    lib().SetConfig.argtypes = [c.c_void_p, c.c_char_p, c.c_int]
    lib().GetConfig.argtypes = [c.c_void_p, c.POINTER(c.c_char_p),
                                c.POINTER(c.c_int)]
    cfg = b'{"slow_search_time": 77, "refresh_interval": 300}'
    assert lib().SetConfig(eng.h, cfg, len(cfg)) == 0
    lib().GetConfig(eng.h, c.byref(out), c.byref(ln))
    got = _json.loads(c.string_at(out, ln.value).decode())
    assert got["slow_search_time"] == 77
    assert got["refresh_interval"] == 300
    # GetDocByDocID(next=1) walks past deleted docs (gamma_api.h:86)
    eng2 = make_engine("/tmp/gamma_docwalk")
    eng2.create_table(64, "FLAT", '{"metric_type": "L2"}')
    for vid in range(10):
        eng2.add_doc(str(vid), base[vid])
    eng2.delete_doc("3")
    eng2.delete_doc("4")
    lib().GetDocByDocID.argtypes = [c.c_void_p, c.c_int, c.c_char,
                                    c.POINTER(c.c_char_p),
                                    c.POINTER(c.c_int)]
    rc = lib().GetDocByDocID(eng2.h, 2, b"\x01", c.byref(out),
                             c.byref(ln))
    assert rc == 0  # next live doc after 2 is 5; payload is a Doc fbs
    rc = lib().GetDocByDocID(eng2.h, 3, b"\x00", c.byref(out),
                             c.byref(ln))
    assert rc != 0  # deleted doc, no next-walk
    eng2.close()


def test_multi_vector_score_range(data):
    """Per-field min/max in a multi-vector request: a doc must pass
    EVERY field's score window to survive the intersection
    (per-field IsSimilarScoreValid, gamma_common_data.h:94)."""
    base, q = data
    n, topn = 3000, 20
    eng = make_engine("/tmp/gamma_mv_range")
    eng.create_table(64, "FLAT", '{"metric_type": "L2"}',
                     extra_vecs=[("emb2", 64)])
    for vid in range(n):
        eng.add_doc(str(vid), base[vid], extra_vecs=[("emb2", base[vid])])
    # identical fields: per-field dists equal; cap field 2's max so the
    # worse half of the top-20 drops out of the intersection
    full = eng.search_pb(q[:2], topn=topn,
                         extra_vec_queries=[("emb2", q[:2])],
                         multi_vector_rank=1)
    for t in range(2):
        scores = [it["score"] for it in full[t]["items"]]
        cut = scores[9]  # combined == per-field dist here
        ranged = eng.search_pb(
            q[t:t + 1], topn=topn,
            extra_vec_queries=[("emb2", q[t:t + 1], None, cut)],
            multi_vector_rank=1)
        rids = [int(it["fields"]["_id"]) for it in ranged[0]["items"]]
        fids = [int(it["fields"]["_id"]) for it in full[t]["items"]]
        assert rids == fids[:10]
    eng.close()


def test_multi_vector_update_delete_rebuild(data):
    """Multi-vector tables under mutation: updating a doc replaces its
    row in EVERY field's index (old rows deleted in each extra index),
    deletes hide the doc from merged results, and RebuildIndex retrains
    the extra fields' indexes together with the primary."""
    import ctypes as c
    from vearch_amd.engine import lib
    base, q = data
    n, topn = 5000, 10
    eng = make_engine("/tmp/gamma_mv_mut")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 4000}',
        extra_vecs=[("emb2", 64)])
    for vid in range(n):
        eng.add_doc(str(vid), base[vid], extra_vecs=[("emb2", base[vid])])
    eng.build_index()

    def top_ids(t=0):
        res = eng.search_pb(q[t:t + 1], topn=topn,
                            index_params='{"recall_num": 50}',
                            extra_vec_queries=[("emb2", q[t:t + 1])],
                            multi_vector_rank=1)
        return [int(it["fields"]["_id"]) for it in res[0]["items"]]

    ids0 = top_ids()
    assert ids0
    victim = ids0[0]
    # update: move the top doc's vectors far away in BOTH fields
    far = base[victim] + 100.0
    eng.add_doc(str(victim), far, extra_vecs=[("emb2", far)])
    ids1 = top_ids()
    assert victim not in ids1
    # delete the next top doc entirely
    victim2 = ids1[0]
    eng.delete_doc(str(victim2))
    ids2 = top_ids()
    assert victim2 not in ids2
    # rebuild from scratch: extras retrain too; merged results stay sane
    lib().RebuildIndex.argtypes = [c.c_void_p, c.c_int, c.c_int, c.c_int]
    assert lib().RebuildIndex(eng.h, 1, 0, 0) == 0
    ids3 = top_ids()
    assert ids3 and victim not in ids3 and victim2 not in ids3
    # identical fields => merged order echoes a single-field search
    single = eng.search_pb(q[:1], topn=topn,
                           index_params='{"recall_num": 50}')
    sids = [int(it["fields"]["_id"]) for it in single[0]["items"]]
    assert ids3 == sids
    eng.close()


def test_lockfree_add_under_search(data):
    """§8f-3 lock-free realtime add: a writer thread appends fresh docs
    (the common pure-append path runs under the SHARED lock — the
    retrieve_idx_pos_ publication pattern, realtime_mem_data.cc:57-68)
    while searcher threads run continuously. Every search must see a
    consistent prefix: valid ids only, every returned id resolves to a
    non-empty pkey and its scalar field, and the observed doc count
    never goes backwards. After the writer joins, all appended docs are
    indexed and searchable."""
    import threading
    from vearch_amd import fbsenc
    base, q = data
    n0, n_new = 4000, 3000
    eng = make_engine("/tmp/gamma_lockfree")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 4000}',
        scalar_fields=[("tag", fbsenc.DATA_STRING)])
    for vid in range(n0):
        eng.add_doc(str(vid), base[vid],
                    fields=[("tag", b"t%d" % (vid % 5),
                             fbsenc.DATA_STRING)])
    eng.build_index()
    done = threading.Event()
    errors = []

    def writer():
        try:
            for vid in range(n0, n0 + n_new):
                eng.add_doc(str(vid), base[vid % len(base)],
                            fields=[("tag", b"t%d" % (vid % 5),
                                     fbsenc.DATA_STRING)])
        except Exception as ex:  # noqa: BLE001
            errors.append(ex)
        finally:
            done.set()

    def searcher(tid):
        try:
            last_count = 0
            while not done.is_set():
                res = eng.search_pb(q[tid * 4:tid * 4 + 4], topn=10,
                                    fields=("_id", "tag"))
                count = eng.num_docs()
                assert count >= last_count  # monotone publication
                last_count = count
                for r in res:
                    for it in r["items"]:
                        vid = int(it["fields"]["_id"])
                        # id valid, row state fully published
                        assert 0 <= vid < n0 + n_new
                        assert it["fields"]["_id"] != ""
                        assert it["fields"]["tag"] == \
                            b"t%d" % (vid % 5)
        except Exception as ex:  # noqa: BLE001
            errors.append(ex)

    threads = [threading.Thread(target=searcher, args=(t,))
               for t in range(3)]
    wt = threading.Thread(target=writer)
    for t in threads:
        t.start()
    wt.start()
    wt.join(timeout=300)
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors
    assert eng.num_docs() == n0 + n_new
    # every appended doc is fetchable and indexed
    docs = eng.query_pb(document_ids=[str(n0 + n_new - 1)])
    assert docs[0]["items"]
    # a query equal to an appended vector finds it (or its duplicate
    # source row — same vector, ties by id): with the exact rerank leg
    # the top hit's canonical distance is exactly 0 at nprobe=nlist
    probe = base[(n0 + 17) % len(base)]
    gd, gi = eng.raw_search(probe[None, :], 3, nprobe=32, rerank=32)
    assert gd[0, 0] == 0.0
    eng.close()


def test_concurrent_searches_deterministic(data):
    """4 threads searching the same batch concurrently (each on its own
    SearchScratch stream) must all return exactly the sequential
    result — concurrency must not perturb selection or rerank."""
    import threading
    base, q = data
    eng = make_engine("/tmp/gamma_conc2")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    want_d, want_i = eng.raw_search(q, 10, nprobe=16, rerank=50)
    results = [None] * 6
    errors = []

    def searcher(tid):
        try:
            results[tid] = eng.raw_search(q, 10, nprobe=16, rerank=50)
        except Exception as ex:  # noqa: BLE001
            errors.append(ex)

    threads = [threading.Thread(target=searcher, args=(t,))
               for t in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors
    for gd, gi in results:
        assert np.array_equal(gi, want_i)
        assert np.array_equal(gd, want_d)
    eng.close()


def test_flat_chunked_gemm_seeded_select_bitexact():
    """FLAT at nq>=512 and n>=200k takes the chunked-GEMM route: one
    seeded select per 65536-vector chunk accumulates the running top-k
    across chunks. With small k (no rerank leg) that select runs on the
    wave-per-query path with seeding — cover it bit-exactly vs the
    oracle (ids AND canonicalized distances)."""
    base = orc.gen_clustered(260000, 32, seed=11, ncl=300)
    q = orc.gen_queries(base, 512, seed=12)
    eng = make_engine("/tmp/gamma_flat_big")
    eng.create_table(32, "FLAT", '{"metric_type": "L2"}')
    eng.add(base)
    gd, gi = eng.raw_search(q, 10, rerank=0)
    od, oi = orc.flat_search(base, q, 10, "L2")
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    # and the IP metric through the same route
    gdi, gii = eng.raw_search(q, 10, rerank=0, metric=2)
    odi, oii = orc.flat_search(base, q, 10, "IP")
    assert np.array_equal(gii, oii)
    assert np.array_equal(gdi, odi)
    eng.close()


def test_opq_train_search_parity(data):
    """OPQ pre-rotation (ivfpq.cc:168-177 params, :362-364 train,
    :585-588 search, :735 raw-space rerank): R is orthonormal, the
    ADC path in rotated space stays bit-exact vs the oracle (fed the
    engine-rotated queries), recall with the raw-space rerank holds the
    reference floors, and dump/load round-trips R."""
    base, q = data
    os.makedirs("/tmp/gamma_opq", exist_ok=True)
    eng = make_engine("/tmp/gamma_opq")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000, "opq": {"nsubvector": 16}}')
    eng.add(base)
    eng.build_index()
    # R orthonormal
    R = eng.debug_opq(64)
    assert np.allclose(R @ R.T, np.eye(64), atol=2e-3)
    # ADC bit-exact in rotated space: oracle gets the engine's own
    # GPU-rotated queries + rotated-space model, probes from the GPU
    q_rot = eng.debug_apply_opq(q)
    gd, gi = eng.raw_search(q, 10, nprobe=16)
    ox = _oracle_from_engine(eng, 64, 64, 16)
    pdists, probes = eng.debug_coarse_assign(q_rot, 16)
    od, oi = ox.search_pct1(q_rot, 10, 16, probes=probes,
                            probe_dists=pdists)
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    # recall floors with the raw-space rerank leg
    _, gti = orc.flat_topk_f64(base, q, 100)
    rd, ri = eng.raw_search(q, 10, nprobe=16, rerank=100)
    assert orc.recall_at(gti, ri, 10) >= 0.9
    # rerank distances are raw-space exact
    lib = RefLib.lib()
    lib.oracle_l2sqr.restype = ctypes.c_float
    for t in range(0, q.shape[0], 7):
        for j in range(3):
            if ri[t, j] < 0:
                continue
            want = lib.oracle_l2sqr(_fp(_c(q[t], np.float32)),
                                    _fp(_c(base[ri[t, j]], np.float32)),
                                    64)
            assert rd[t, j] == want
    # dump/load: identical results, R survives
    eng.dump()
    eng.close()
    eng2 = make_engine("/tmp/gamma_opq")
    eng2.create_table(
        64, "IVFPQ",
        '{"ncentroids": 64, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 8000, "opq": {"nsubvector": 16}}')
    eng2.load()
    R2 = eng2.debug_opq(64)
    assert np.array_equal(R, R2)
    gd2, gi2 = eng2.raw_search(q, 10, nprobe=16)
    assert np.array_equal(gd, gd2) and np.array_equal(gi, gi2)
    eng2.close()


def test_ivfpq_d768_m96_bitexact():
    """Config-5 shape (d=768, m=96, dsub=8): the wide-M template path
    stays bit-exact vs the oracle."""
    base = orc.gen_clustered(12000, 768, seed=9, ncl=64)
    q = orc.gen_queries(base, 16, seed=10)
    eng = make_engine("/tmp/gamma_d768")
    eng.create_table(
        768, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 96, "metric_type": "L2", '
        '"training_threshold": 6000}')
    eng.add(base)
    eng.build_index()
    gd, gi = eng.raw_search(q, 10, nprobe=8)
    ox = orc.OracleIVFPQ(768, 32, 96, metric="L2")
    cent, books = eng.debug_model(32, 768, 96)
    ox.centroids, ox.codebooks = cent, books
    ids_all, codes_all, offsets = [], [], [0]
    for ln in range(32):
        li, lc = eng.debug_list(ln, 96)
        ids_all.append(li)
        codes_all.append(lc)
        offsets.append(offsets[-1] + len(li))
    ox.ids = np.concatenate(ids_all)
    ox.codes = np.concatenate(codes_all)
    ox.offsets = np.array(offsets, dtype=np.int64)
    pdists, probes = eng.debug_coarse_assign(q, 8)
    od, oi = ox.search_pct1(q, 10, 8, probes=probes, probe_dists=pdists)
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    eng.close()


def test_ivfflat_ip_bitexact(data):
    """IVFFLAT + InnerProduct (spherical k-means assign, descending
    similarity) — canonical re-rank makes returned scores bit-exact."""
    base, q = data
    eng = make_engine("/tmp/gamma_ivfflat_ip")
    eng.create_table(
        64, "IVFFLAT",
        '{"ncentroids": 64, "metric_type": "InnerProduct", '
        '"training_threshold": 8000}')
    eng.add(base)
    eng.build_index()
    gd, gi = eng.raw_search(q, 10, nprobe=16)
    cent, _ = eng.debug_model(64, 64, 0)
    ids_all, vec_all, offsets = [], [], [0]
    for ln in range(64):
        li, lc = eng.debug_list(ln, 64 * 4)
        ids_all.append(li)
        vec_all.append(lc.view(np.float32).reshape(-1, 64))
        offsets.append(offsets[-1] + len(li))
    ids = np.concatenate(ids_all)
    vecs = np.ascontiguousarray(np.concatenate(vec_all), dtype=np.float32)
    offsets = np.array(offsets, dtype=np.int64)
    _, probes = eng.debug_coarse_assign(q, 16)
    from oracle.gamma_oracle import _ip64, _up8
    lib = RefLib.lib()
    od = np.empty((64, 10), dtype=np.float32)
    oi = np.empty((64, 10), dtype=np.int64)
    lib.oracle_ivfflat_search(
        64, 64, 64, _fp(_c(q, np.float32)), _ip64(offsets), _ip64(ids),
        _fp(vecs), 16, _ip64(_c(probes, np.int64)), _up8(None), 1, 10,
        _fp(od), _ip64(oi))
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    for t in range(q.shape[0]):
        row = [gd[t, j] for j in range(10) if gi[t, j] >= 0]
        assert row == sorted(row, reverse=True)
    eng.close()


def test_flat_gemm_path_ip_bitexact():
    """IP ordering through the chunked GEMM + seeded select path
    (nq >= 512, N >= 200k)."""
    base = orc.gen_clustered(210000, 32, seed=15, ncl=300)
    q = orc.gen_queries(base, 520, seed=16)
    eng = make_engine("/tmp/gamma_flat_gemm_ip")
    eng.create_table(32, "FLAT", '{"metric_type": "InnerProduct"}')
    eng.add(base)
    gd, gi = eng.raw_search(q, 10)
    od, oi = orc.flat_search(base, q, 10, "InnerProduct")
    assert np.array_equal(gi, oi)
    assert np.array_equal(gd, od)
    eng.close()


def test_cached_queries_equivalent(data, ivfpq_engine):
    """GammaCacheQueries + GammaRawSearchCached (the bench hot path) must
    equal the host-pointer search exactly."""
    base, q = data
    eng = ivfpq_engine
    gd0, gi0 = eng.raw_search(q, 10, nprobe=16, rerank=100)
    nq = eng.cache_queries(q)
    gd1, gi1 = eng.search_cached(nq, 10, nprobe=16, rerank=100)
    assert np.array_equal(gi0, gi1)
    assert np.array_equal(gd0, gd1)


def test_metric_override_per_query(data, ivfpq_engine):
    """Per-request metric override (retrieval params metric_type,
    ivfpq.cc:247-259): IP search on an L2-trained index runs the IP
    table path and orders descending."""
    base, q = data
    eng = ivfpq_engine
    gd, gi = eng.raw_search(q[:8], 5, nprobe=16, metric=2)
    for t in range(8):
        row = [gd[t, j] for j in range(5) if gi[t, j] >= 0]
        assert row == sorted(row, reverse=True)
    # exact IP values after canonical rerank? no rerank requested ->
    # ADC-IP values; check ordering + plausibility vs exact IP top-1
    odf, oif = orc.flat_search(base, q[:8], 50, "InnerProduct")
    overlap = np.mean([len(set(gi[t].tolist()) & set(oif[t].tolist())) / 5
                       for t in range(8)])
    assert overlap >= 0.4, overlap


def test_search_deterministic_across_runs(data, ivfpq_engine):
    """The selector's LDS append order is nondeterministic; the exact
    (dist,id) total order must make results run-invariant anyway."""
    base, q = data
    eng = ivfpq_engine
    runs = [eng.raw_search(q, 10, nprobe=16, rerank=100)
            for _ in range(3)]
    for gd, gi in runs[1:]:
        assert np.array_equal(gi, runs[0][1])
        assert np.array_equal(gd, runs[0][0])
    runs = [eng.raw_search(q, 10, nprobe=16) for _ in range(3)]
    for gd, gi in runs[1:]:
        assert np.array_equal(gi, runs[0][1])
        assert np.array_equal(gd, runs[0][0])


def test_kill_mid_flight_machinery(data, ivfpq_engine):
    """SetKillStatus during an in-flight search: the armed device flag
    stops the scan between lists (ivfpq.h:927 analog). Timing-dependent,
    so accept either a killed or a completed search — but the engine
    must stay healthy afterwards."""
    import threading
    from vearch_amd import clear_kill, set_kill
    base, q = data
    eng = ivfpq_engine
    outcome = {}

    def searcher():
        try:
            outcome["res"] = eng.search_pb(q, topn=10,
                                           index_params='{"nprobe": 64}',
                                           request_id="midkill",
                                           partition_id=2)
        except InterruptedError:
            outcome["killed"] = True
        except RuntimeError as ex:
            outcome["err"] = ex

    t = threading.Thread(target=searcher)
    t.start()
    set_kill("midkill", 2)
    t.join(timeout=60)
    clear_kill("midkill", 2)
    assert "err" not in outcome, outcome
    # engine healthy afterwards
    res = eng.search_pb(q[:4], topn=5, index_params='{"nprobe": 16}')
    assert len(res) == 4 and res[0]["items"]


def test_rebuild_index(data):
    """RebuildIndex(drop=1) retrains and re-adds everything; results stay
    parity-clean afterwards."""
    import ctypes as c
    from vearch_amd.engine import lib
    base, q = data
    eng = make_engine("/tmp/gamma_rebuild")
    eng.create_table(
        64, "IVFPQ",
        '{"ncentroids": 32, "nsubvector": 16, "metric_type": "L2", '
        '"training_threshold": 4000}')
    eng.add(base[:9000])
    eng.build_index()
    gd0, gi0 = eng.raw_search(q, 10, nprobe=32, rerank=100)
    # grow the corpus, then rebuild from scratch
    eng.add(base[9000:12000])
    lib().RebuildIndex.argtypes = [c.c_void_p, c.c_int, c.c_int, c.c_int]
    assert lib().RebuildIndex(eng.h, 1, 0, 0) == 0
    gd1, gi1 = eng.raw_search(q, 10, nprobe=32, rerank=100)
    # rebuilt model differs, but recall must hold on the larger corpus
    _, gti = orc.flat_topk_f64(base[:12000], q, 10)
    assert orc.recall_at(gti, gi1, 10) >= 0.9
    # and the total indexed count covers everything
    total = sum(len(eng.debug_list(ln, 16)[0]) for ln in range(32))
    assert total == 12000
    eng.close()
