"""CPU oracle for the Gamma hot path.

Pure-numpy restatement of the training algorithms (k-means, PQ) plus a
ctypes wrapper over oracle/ref_scan.c, whose sequential-fmaf loops define
the canonical fp32 arithmetic the HIP kernels are held bit-exact to.

Reference anchors (/root/reference):
  - k-means: faiss Clustering as configured by gamma_index_ivfpq.cc:188-191
    (cp.niter = 10, spherical for inner product), Lloyd iterations with
    random-sample init and largest-cluster split for empty clusters —
    faiss's published algorithm; bit-exact parity with faiss is UNPINNED
    (faiss v1.14.1 is an external, un-vendored dependency).
  - PQ train: faiss ProductQuantizer::train — per-subspace k-means with
    ksub = 2^nbits centroids (niter 25, faiss default).
  - search semantics: see oracle/ref_scan.c header.

TEST INFRASTRUCTURE ONLY — never part of the product path.
"""
import ctypes
import os
import subprocess

import numpy as np

_ORACLE_DIR = os.path.dirname(os.path.abspath(__file__))


# ---------------------------------------------------------------- C library
class RefLib:
    """ctypes wrapper over oracle/_ref/libgammaoracle.so (built by
    `make -C oracle`, auto-built on first use if gcc is available)."""

    _lib = None

    @classmethod
    def lib(cls):
        if cls._lib is None:
            so = os.path.join(_ORACLE_DIR, "_ref", "libgammaoracle.so")
            if not os.path.exists(so):
                subprocess.check_call(["make", "-C", _ORACLE_DIR, "-s"])
            lib = ctypes.CDLL(so)
            f32p = ctypes.POINTER(ctypes.c_float)
            i64p = ctypes.POINTER(ctypes.c_int64)
            u8p = ctypes.POINTER(ctypes.c_uint8)
            lib.oracle_flat_search.argtypes = [
                ctypes.c_int64, ctypes.c_int, f32p, ctypes.c_int, f32p,
                ctypes.c_int, u8p, ctypes.c_int, f32p, i64p]
            lib.oracle_ivfpq_search.argtypes = [
                ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
                ctypes.c_int, f32p, f32p, f32p, i64p, i64p, u8p,
                ctypes.c_int, i64p, u8p, ctypes.c_int, ctypes.c_int,
                f32p, i64p]
            lib.oracle_ivfflat_search.argtypes = [
                ctypes.c_int, ctypes.c_int, ctypes.c_int, f32p, i64p, i64p,
                f32p, ctypes.c_int, i64p, u8p, ctypes.c_int, ctypes.c_int,
                f32p, i64p]
            lib.oracle_coarse_assign.argtypes = [
                ctypes.c_int, ctypes.c_int, ctypes.c_int, f32p, f32p,
                ctypes.c_int, ctypes.c_int, f32p, i64p]
            lib.oracle_pq_encode.argtypes = [
                ctypes.c_int64, ctypes.c_int, ctypes.c_int, ctypes.c_int,
                f32p, f32p, u8p]
            lib.oracle_adc_table_l2.argtypes = [
                ctypes.c_int, ctypes.c_int, ctypes.c_int, f32p, f32p, f32p,
                f32p]
            lib.oracle_adc_table_ip.argtypes = [
                ctypes.c_int, ctypes.c_int, ctypes.c_int, f32p, f32p, f32p]
            lib.oracle_ivfpq_search_pct1.argtypes = [
                ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
                ctypes.c_int, f32p, f32p, f32p, i64p, i64p, u8p,
                ctypes.c_int, i64p, f32p, u8p, ctypes.c_int, f32p, i64p]
            lib.oracle_l2_gemm_form.restype = ctypes.c_float
            lib.oracle_l2_gemm_form.argtypes = [f32p, f32p, ctypes.c_int]
            lib.oracle_num_threads.restype = ctypes.c_int
            cls._lib = lib
        return cls._lib


def _fp(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def _ip64(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


def _up8(a):
    if a is None:
        return ctypes.cast(None, ctypes.POINTER(ctypes.c_uint8))
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))


def _c(a, dtype):
    return np.ascontiguousarray(a, dtype=dtype)


# ------------------------------------------------------------ data synthesis
def gen_clustered(n, d, seed=42, ncl=1024, sigma=0.10, dtype=np.float32):
    """Clustered-Gaussian synthetic DB (BASELINE.md protocol): ncl true
    centers uniform in [0,1)^d, points = center + N(0, sigma^2)."""
    rng = np.random.default_rng(seed)
    centers = rng.random((ncl, d), dtype=np.float32)
    asg = rng.integers(0, ncl, size=n)
    x = centers[asg] + sigma * rng.standard_normal((n, d), dtype=np.float32)
    return x.astype(dtype)


def gen_queries(base, nq, seed=43, sigma=0.05):
    """Queries = perturbed DB samples."""
    rng = np.random.default_rng(seed)
    idx = rng.integers(0, base.shape[0], size=nq)
    q = base[idx] + sigma * rng.standard_normal(
        (nq, base.shape[1]), dtype=np.float32)
    return q.astype(np.float32)


# ---------------------------------------------------------------- fp64 truth
def flat_topk_f64(base, queries, k, metric="L2", block=2048):
    """Exact fp64 ground truth (recall denominators). Ties by (dist, id)."""
    n = base.shape[0]
    b64 = base.astype(np.float64)
    out_ids = np.empty((queries.shape[0], k), dtype=np.int64)
    out_d = np.empty((queries.shape[0], k), dtype=np.float64)
    for s in range(0, queries.shape[0], block):
        q = queries[s:s + block].astype(np.float64)
        if metric == "L2":
            d2 = (
                (q * q).sum(1)[:, None]
                - 2.0 * (q @ b64.T)
                + (b64 * b64).sum(1)[None, :]
            )
            key = d2
        else:
            key = -(q @ b64.T)
        part = np.argpartition(key, min(k, n - 1), axis=1)[:, :k]
        pk = np.take_along_axis(key, part, 1)
        order = np.lexsort((part, pk), axis=1)
        ids = np.take_along_axis(part, order, 1)
        out_ids[s:s + block] = ids
        kd = np.take_along_axis(pk, order, 1)
        out_d[s:s + block] = kd if metric == "L2" else -kd
    return out_d, out_ids


def recall_at(gt_ids, ids, k):
    """recall@k: |gt_top_k ∩ returned_top_k| / k, averaged over queries
    (the reference cluster tests' definition, test_vector_index_ivfpq.py)."""
    hits = 0
    for g, r in zip(gt_ids[:, :k], ids[:, :k]):
        hits += len(set(g.tolist()) & set(r[r >= 0].tolist()))
    return hits / (gt_ids.shape[0] * k)


# ------------------------------------------------------------------- k-means
def kmeans(x, ncl, niter=10, seed=42, spherical=False):
    """Lloyd k-means, faiss-Clustering-style (cp.niter=10 per
    gamma_index_ivfpq.cc:188; spherical normalization for IP per :190).
    Init: random sample without replacement. Empty clusters: split the
    largest (faiss's documented policy)."""
    x = _c(x, np.float32)
    n, d = x.shape
    rng = np.random.default_rng(seed)
    cent = x[rng.choice(n, size=ncl, replace=(n < ncl))].copy()
    if spherical:
        cent /= np.maximum(np.linalg.norm(cent, axis=1, keepdims=True), 1e-20)
    for _ in range(niter):
        asg = assign_np(x, cent, spherical)
        sums = np.zeros((ncl, d), dtype=np.float64)
        np.add.at(sums, asg, x.astype(np.float64))
        counts = np.bincount(asg, minlength=ncl)
        for c in np.where(counts == 0)[0]:
            big = int(np.argmax(counts))
            cent_big = sums[big] / counts[big]
            eps = 1e-5 * (1.0 + np.abs(cent_big))
            sums[c] = (cent_big + eps) * (counts[big] // 2)
            sums[big] = (cent_big - eps) * (counts[big] - counts[big] // 2)
            counts[c] = counts[big] // 2
            counts[big] -= counts[c]
        cent = (sums / np.maximum(counts, 1)[:, None]).astype(np.float32)
        if spherical:
            cent /= np.maximum(
                np.linalg.norm(cent, axis=1, keepdims=True), 1e-20)
    return cent


def assign_np(x, cent, spherical=False, block=65536):
    """Nearest centroid (L2; spherical uses max dot). Vectorized fp32."""
    out = np.empty(x.shape[0], dtype=np.int64)
    cn = (cent.astype(np.float64) ** 2).sum(1)
    for s in range(0, x.shape[0], block):
        xb = x[s:s + block].astype(np.float64)
        dots = xb @ cent.astype(np.float64).T
        if spherical:
            out[s:s + block] = np.argmax(dots, 1)
        else:
            out[s:s + block] = np.argmin(cn[None, :] - 2 * dots, 1)
    return out


def pq_train(x, M, ksub=256, niter=25, seed=42):
    """Per-subspace k-means (faiss ProductQuantizer::train).
    Returns codebooks (M, ksub, dsub) fp32."""
    n, d = x.shape
    dsub = d // M
    books = np.empty((M, ksub, dsub), dtype=np.float32)
    for m in range(M):
        sub = _c(x[:, m * dsub:(m + 1) * dsub], np.float32)
        books[m] = kmeans(sub, ksub, niter=niter, seed=seed + m)
    return books


# ------------------------------------------------------- full CPU pipeline
class OracleIVFPQ:
    """CPU IVFPQ mirroring GammaIVFPQIndex semantics (by_residual=True,
    use_precomputed_table=0, nbits=8 — gamma_index_ivfpq.cc:195-197).
    Used to pin the GPU engine stage by stage."""

    def __init__(self, d, nlist, M, metric="L2"):
        assert d % M == 0
        self.d, self.nlist, self.M, self.ksub = d, nlist, M, 256
        self.metric = metric
        self.centroids = None
        self.codebooks = None
        self.ids = None          # flat int64, bit 63 = deleted
        self.codes = None        # flat (ntotal, M) u8
        self.offsets = None      # nlist+1

    @property
    def metric_ip(self):
        m = self.metric.lower()
        if m in ("innerproduct", "ip"):
            return 1
        if m == "l2":
            return 0
        raise ValueError(f"unknown metric: {self.metric}")

    def train(self, xt, seed=42):
        spherical = self.metric_ip == 1
        self.centroids = kmeans(xt, self.nlist, niter=10, seed=seed,
                                spherical=spherical)
        asg = assign_np(xt, self.centroids, spherical)
        resid = xt - self.centroids[asg]
        self.codebooks = pq_train(resid, self.M, self.ksub, niter=25,
                                  seed=seed)

    def add(self, x, base_id=0):
        spherical = self.metric_ip == 1
        asg = assign_np(x, self.centroids, spherical)
        resid = _c(x - self.centroids[asg], np.float32)
        codes = np.empty((x.shape[0], self.M), dtype=np.uint8)
        RefLib.lib().oracle_pq_encode(
            x.shape[0], self.d, self.M, self.ksub, _fp(resid),
            _fp(_c(self.codebooks, np.float32)), _up8(codes))
        ids = np.arange(base_id, base_id + x.shape[0], dtype=np.int64)
        order = np.argsort(asg, kind="stable")
        self.ids = ids[order].copy()
        self.codes = codes[order].copy()
        counts = np.bincount(asg, minlength=self.nlist)
        self.offsets = np.zeros(self.nlist + 1, dtype=np.int64)
        np.cumsum(counts, out=self.offsets[1:])
        self.assignments = asg
        return asg

    def coarse_assign(self, q, nprobe):
        q = _c(q, np.float32)
        nq = q.shape[0]
        dists = np.empty((nq, nprobe), dtype=np.float32)
        lists = np.empty((nq, nprobe), dtype=np.int64)
        RefLib.lib().oracle_coarse_assign(
            nq, self.d, self.nlist, _fp(q),
            _fp(_c(self.centroids, np.float32)), nprobe, self.metric_ip,
            _fp(dists), _ip64(lists))
        return dists, lists

    def coarse_gemm_dists(self, q, probes):
        """GEMM-form coarse distances fmaf(-2,dot,qn+cn) — the engine's
        probe-distance arithmetic, canonical fmaf order."""
        lib = RefLib.lib()
        q = _c(q, np.float32)
        cent = _c(self.centroids, np.float32)
        out = np.empty(probes.shape, dtype=np.float32)
        for i in range(q.shape[0]):
            for p in range(probes.shape[1]):
                ln = probes[i, p]
                if ln < 0:
                    out[i, p] = -1.0
                    continue
                out[i, p] = lib.oracle_l2_gemm_form(
                    _fp(q[i]), _fp(cent[ln]), self.d)
        return out

    def search_pct1(self, q, k, nprobe, probes=None, probe_dists=None,
                    del_bitmap=None):
        """L2 search with decomposed tables (use_precomputed_table=1
        semantics) — the mode the HIP engine runs; bit-exact vs the GPU
        given the same probes + probe_dists."""
        assert self.metric_ip == 0
        q = _c(q, np.float32)
        nq = q.shape[0]
        if probes is None:
            _, probes = self.coarse_assign(q, nprobe)
        probes = _c(probes, np.int64)
        if probe_dists is None:
            probe_dists = self.coarse_gemm_dists(q, probes)
        probe_dists = _c(probe_dists, np.float32)
        dists = np.empty((nq, k), dtype=np.float32)
        ids = np.empty((nq, k), dtype=np.int64)
        RefLib.lib().oracle_ivfpq_search_pct1(
            nq, self.d, self.M, self.ksub, self.nlist, _fp(q),
            _fp(_c(self.centroids, np.float32)),
            _fp(_c(self.codebooks, np.float32)),
            _ip64(_c(self.offsets, np.int64)), _ip64(_c(self.ids, np.int64)),
            _up8(_c(self.codes, np.uint8)), nprobe, _ip64(probes),
            _fp(probe_dists), _up8(del_bitmap), k, _fp(dists), _ip64(ids))
        return dists, ids

    def search(self, q, k, nprobe, probes=None, del_bitmap=None):
        q = _c(q, np.float32)
        nq = q.shape[0]
        if probes is None:
            _, probes = self.coarse_assign(q, nprobe)
        probes = _c(probes, np.int64)
        dists = np.empty((nq, k), dtype=np.float32)
        ids = np.empty((nq, k), dtype=np.int64)
        RefLib.lib().oracle_ivfpq_search(
            nq, self.d, self.M, self.ksub, self.nlist, _fp(q),
            _fp(_c(self.centroids, np.float32)),
            _fp(_c(self.codebooks, np.float32)),
            _ip64(_c(self.offsets, np.int64)), _ip64(_c(self.ids, np.int64)),
            _up8(_c(self.codes, np.uint8)), nprobe, _ip64(probes),
            _up8(del_bitmap), self.metric_ip, k, _fp(dists), _ip64(ids))
        return dists, ids


def flat_search(base, queries, k, metric="L2", del_bitmap=None):
    """Canonical fp32 FLAT search (ties by (dist,id)) via the C oracle."""
    base = _c(base, np.float32)
    queries = _c(queries, np.float32)
    nq = queries.shape[0]
    dists = np.empty((nq, k), dtype=np.float32)
    ids = np.empty((nq, k), dtype=np.int64)
    if metric in ("InnerProduct", "IP"):
        mip = 1
    elif metric == "L2":
        mip = 0
    else:
        raise ValueError(f"unknown metric {metric!r}")
    RefLib.lib().oracle_flat_search(
        base.shape[0], base.shape[1], _fp(base), nq, _fp(queries), k,
        _up8(del_bitmap), mip, _fp(dists), _ip64(ids))
    return dists, ids
