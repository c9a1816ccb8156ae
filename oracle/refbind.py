"""ctypes bindings for oracle/_ref/libgammaref.so — the REFERENCE'S OWN
scanner code compiled against the faiss stub (see ref_harness.cpp /
ref_extract.sh). TEST INFRASTRUCTURE ONLY: only tests and the golden
fixture generator may import this; nothing in the product path loads it.

The library can only be BUILT where /root/reference exists, but the
built .so travels with the repo snapshot, so tests that find it present
run anywhere.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "_ref", "libgammaref.so")


def available():
    return os.path.exists(_SO)


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = ctypes.CDLL(_SO)
        i64p = np.ctypeslib.ndpointer(np.int64, flags="C")
        f32p = np.ctypeslib.ndpointer(np.float32, flags="C")
        u8p = np.ctypeslib.ndpointer(np.uint8, flags="C")
        _lib.ref_flat_search.restype = ctypes.c_int
        _lib.ref_flat_search.argtypes = [
            ctypes.c_int64, ctypes.c_int, f32p, ctypes.c_void_p,
            ctypes.c_int, f32p, ctypes.c_int, ctypes.c_int, f32p, i64p]
        _lib.ref_ivfpq_search.restype = ctypes.c_int
        _lib.ref_ivfpq_search.argtypes = [
            ctypes.c_int, ctypes.c_int, ctypes.c_int, f32p, f32p, i64p,
            i64p, u8p, ctypes.c_int, f32p, ctypes.c_int, i64p, f32p,
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
            f32p, i64p]
        _lib.ref_ivfflat_search.restype = ctypes.c_int
        _lib.ref_ivfflat_search.argtypes = [
            ctypes.c_int, ctypes.c_int, i64p, i64p, f32p, ctypes.c_int,
            f32p, ctypes.c_int, i64p, ctypes.c_int, ctypes.c_int,
            ctypes.c_void_p, f32p, i64p]
        _lib.ref_ivfpq_precomputed_table.restype = ctypes.c_int
        _lib.ref_ivfpq_precomputed_table.argtypes = [
            ctypes.c_int, ctypes.c_int, ctypes.c_int, f32p, f32p, f32p]
    return _lib


def _bm(del_bitmap):
    if del_bitmap is None:
        return None
    return del_bitmap.ctypes.data_as(ctypes.c_void_p)


def flat_search(base, queries, k, metric="L2", del_bitmap=None):
    base = np.ascontiguousarray(base, np.float32)
    q = np.ascontiguousarray(queries, np.float32)
    nq = q.shape[0]
    out_d = np.empty((nq, k), np.float32)
    out_i = np.empty((nq, k), np.int64)
    rc = lib().ref_flat_search(base.shape[0], base.shape[1], base,
                               _bm(del_bitmap), nq, q, k,
                               1 if metric == "IP" else 0, out_d, out_i)
    assert rc == 0
    return out_d, out_i


def ivfpq_search(ox, queries, k, nprobe, probes, probe_dists,
                 use_precomputed_table, metric="L2", del_bitmap=None):
    """Run the reference scanner on an OracleIVFPQ's model + lists."""
    q = np.ascontiguousarray(queries, np.float32)
    nq = q.shape[0]
    out_d = np.empty((nq, k), np.float32)
    out_i = np.empty((nq, k), np.int64)
    rc = lib().ref_ivfpq_search(
        ox.d, ox.nlist, ox.M,
        np.ascontiguousarray(ox.centroids, np.float32),
        np.ascontiguousarray(ox.codebooks, np.float32),
        np.ascontiguousarray(ox.offsets, np.int64),
        np.ascontiguousarray(ox.ids, np.int64),
        np.ascontiguousarray(ox.codes, np.uint8), nq, q, nprobe,
        np.ascontiguousarray(probes, np.int64),
        np.ascontiguousarray(probe_dists, np.float32),
        1 if metric == "IP" else 0, use_precomputed_table, k,
        _bm(del_bitmap), out_d, out_i)
    assert rc == 0
    return out_d, out_i


def ivfflat_search(d, nlist, offsets, ids, vecs, queries, k, nprobe,
                   probes, metric="L2", del_bitmap=None):
    q = np.ascontiguousarray(queries, np.float32)
    nq = q.shape[0]
    out_d = np.empty((nq, k), np.float32)
    out_i = np.empty((nq, k), np.int64)
    rc = lib().ref_ivfflat_search(
        d, nlist, np.ascontiguousarray(offsets, np.int64),
        np.ascontiguousarray(ids, np.int64),
        np.ascontiguousarray(vecs, np.float32), nq, q, nprobe,
        np.ascontiguousarray(probes, np.int64),
        1 if metric == "IP" else 0, k, _bm(del_bitmap), out_d, out_i)
    assert rc == 0
    return out_d, out_i
