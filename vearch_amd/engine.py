"""ctypes host wrapper over libgamma.so — plays the role of the Go cgo
binding (reference: internal/engine/sdk/go/gamma/gamma.go). The product
compute path is entirely inside libgamma.so (HIP, gfx950); this wrapper
only marshals buffers. It FAILS LOUDLY if the native library is missing —
there is no CPU fallback."""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_DIR, "libgamma.so")


class CStatus(ctypes.Structure):
    _fields_ = [("code", ctypes.c_int), ("msg", ctypes.c_char_p)]


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIB_PATH):
            raise RuntimeError(
                f"libgamma.so not found at {LIB_PATH}: build it with "
                "__graft_entry__.build() — the MI355X engine has no "
                "CPU fallback")
        L = ctypes.CDLL(LIB_PATH)
        c = ctypes
        f32p = c.POINTER(c.c_float)
        i64p = c.POINTER(c.c_int64)
        u8p = c.POINTER(c.c_uint8)
        L.Init.restype = c.c_void_p
        L.Init.argtypes = [c.c_char_p, c.c_int]
        L.Close.argtypes = [c.c_void_p]
        L.CreateTable.restype = CStatus
        L.CreateTable.argtypes = [c.c_void_p, c.c_char_p, c.c_int]
        L.AddOrUpdateDoc.argtypes = [c.c_void_p, c.c_char_p, c.c_int]
        L.DeleteDoc.argtypes = [c.c_void_p, c.c_char_p, c.c_int]
        L.BuildIndex.argtypes = [c.c_void_p]
        L.Dump.argtypes = [c.c_void_p]
        L.Load.argtypes = [c.c_void_p]
        L.Search.restype = CStatus
        L.Search.argtypes = [c.c_void_p, c.c_char_p, c.c_int,
                             c.POINTER(c.c_char_p), c.POINTER(c.c_int)]
        L.Query.restype = CStatus
        L.Query.argtypes = [c.c_void_p, c.c_char_p, c.c_int,
                            c.POINTER(c.c_char_p), c.POINTER(c.c_int)]
        L.GetEngineStatus.argtypes = [c.c_void_p, c.POINTER(c.c_char_p),
                                      c.POINTER(c.c_int)]
        L.GetDocByID.argtypes = [c.c_void_p, c.c_char_p, c.c_int,
                                 c.POINTER(c.c_char_p), c.POINTER(c.c_int)]
        L.SetKillStatus.argtypes = [c.c_char_p, c.c_int, c.c_int]
        L.DeleteKillStatus.argtypes = [c.c_char_p, c.c_int]
        L.Backup.restype = CStatus
        L.Backup.argtypes = [c.c_void_p, c.c_int]
        L.AddFieldIndexWithParams.restype = CStatus
        L.AddFieldIndexWithParams.argtypes = [
            c.c_void_p, c.c_char_p, c.c_int,
            c.POINTER(c.c_char_p), c.POINTER(c.c_int), c.c_int,
            c.c_char_p, c.c_int, c.c_char_p, c.c_int]
        L.RemoveFieldIndex.restype = CStatus
        L.RemoveFieldIndex.argtypes = [c.c_void_p, c.c_char_p, c.c_int]
        L.GammaBulkAdd.argtypes = [c.c_void_p, c.c_char_p, c.c_int, c.c_int,
                                   f32p]
        L.GammaRawSearch.argtypes = [c.c_void_p, c.c_int, f32p, c.c_int,
                                     c.c_int, c.c_int, c.c_int, f32p, i64p]
        L.GammaCacheQueries.argtypes = [c.c_void_p, c.c_int, f32p]
        L.GammaRawSearchCached.argtypes = [c.c_void_p, c.c_int, c.c_int,
                                           c.c_int, c.c_int, c.c_int, f32p,
                                           i64p]
        L.GammaDebugCoarseAssign.argtypes = [c.c_void_p, c.c_int, f32p,
                                             c.c_int, i64p, f32p]
        L.GammaDebugGetModel.argtypes = [c.c_void_p, f32p, f32p]
        L.GammaDebugGetOPQ.argtypes = [c.c_void_p, f32p]
        L.GammaDebugApplyOPQ.argtypes = [c.c_void_p, f32p, c.c_int, f32p]
        L.GammaDebugGetList.restype = c.c_int64
        L.GammaDebugGetList.argtypes = [c.c_void_p, c.c_int64, i64p, u8p]
        L.GammaDebugNumDocs.restype = c.c_int64
        L.GammaDebugNumDocs.argtypes = [c.c_void_p]
        L.GammaLastSearchTiming.argtypes = [c.c_void_p,
                                            c.POINTER(c.c_double)]
        _lib = L
    return _lib


def _fp(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def _ip(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


def _check(st: CStatus, what):
    if st.code != 0:
        msg = st.msg.decode() if st.msg else ""
        raise RuntimeError(f"{what} failed (code {st.code}): {msg}")


class GammaEngine:
    """One Gamma engine == one Vearch partition on one GPU."""

    def __init__(self, path=".", space_name="bench_space"):
        cfg = ('{"path": "%s", "log_dir": "%s", "space_name": "%s"}'
               % (path, path, space_name))
        self.h = lib().Init(cfg.encode(), len(cfg))
        if not self.h:
            raise RuntimeError("Init failed (GPU required)")
        self.d = 0
        self.vec_name = "emb"

    def close(self):
        if self.h:
            lib().Close(self.h)
            self.h = None

    def create_table(self, d, index_type="IVFPQ", index_params="",
                     name="bench_space", scalar_fields=(), vec_name="emb",
                     extra_vecs=()):
        """extra_vecs: additional (name, dim) vector fields of a
        multi-vector table (vector_manager.cc multi-field dispatch)."""
        from . import fbsenc
        self.d = d
        self.vec_name = vec_name
        buf = fbsenc.build_table(name, list(scalar_fields), vec_name, d,
                                 index_type, index_params,
                                 extra_vecs=extra_vecs)
        _check(lib().CreateTable(self.h, buf, len(buf)), "CreateTable")

    def add(self, vecs):
        vecs = np.ascontiguousarray(vecs, dtype=np.float32)
        rc = lib().GammaBulkAdd(self.h, self.vec_name.encode(),
                                len(self.vec_name), vecs.shape[0],
                                _fp(vecs))
        if rc != 0:
            raise RuntimeError(f"GammaBulkAdd failed rc={rc}")

    def add_doc(self, p_key, vec, fields=(), extra_vecs=()):
        """extra_vecs: [(name, array)] — one entry per extra vector
        field of a multi-vector table."""
        from . import fbsenc
        vec = np.ascontiguousarray(vec, dtype=np.float32)
        fl = [("_id", p_key.encode(), fbsenc.DATA_STRING)]
        for name, value, dt in fields:
            fl.append((name, value, dt))
        fl.append((self.vec_name, vec.tobytes(), fbsenc.DATA_VECTOR))
        for name, v in extra_vecs:
            v = np.ascontiguousarray(v, dtype=np.float32)
            fl.append((name, v.tobytes(), fbsenc.DATA_VECTOR))
        buf = fbsenc.build_doc(fl)
        rc = lib().AddOrUpdateDoc(self.h, buf, len(buf))
        if rc != 0:
            raise RuntimeError(f"AddOrUpdateDoc failed rc={rc}")

    def delete_doc(self, p_key):
        return lib().DeleteDoc(self.h, p_key.encode(), len(p_key))

    def build_index(self):
        rc = lib().BuildIndex(self.h)
        if rc != 0:
            raise RuntimeError(f"BuildIndex failed rc={rc}")

    def dump(self):
        rc = lib().Dump(self.h)
        if rc != 0:
            raise RuntimeError("Dump failed")

    def load(self):
        rc = lib().Load(self.h)
        if rc != 0:
            raise RuntimeError("Load failed")

    def backup(self, command=0):
        """Backup (gamma_api.h:104): command 0 = create."""
        _check(lib().Backup(self.h, command), "Backup")

    def add_field_index(self, name, fields, index_type="SCALAR",
                        params=""):
        """AddFieldIndexWithParams (gamma_api.h:107)."""
        arr = (ctypes.c_char_p * len(fields))(
            *[f.encode() for f in fields])
        lens = (ctypes.c_int * len(fields))(
            *[len(f.encode()) for f in fields])
        st = lib().AddFieldIndexWithParams(
            self.h, name.encode(), len(name.encode()), arr, lens,
            len(fields), index_type.encode(), len(index_type.encode()),
            params.encode(), len(params.encode()))
        _check(st, "AddFieldIndexWithParams")

    def remove_field_index(self, name):
        """RemoveFieldIndex (gamma_api.h:114)."""
        _check(lib().RemoveFieldIndex(self.h, name.encode(),
                                      len(name.encode())),
               "RemoveFieldIndex")

    def num_docs(self):
        return lib().GammaDebugNumDocs(self.h)

    def status(self):
        out = ctypes.c_char_p()
        n = ctypes.c_int()
        lib().GetEngineStatus(self.h, ctypes.byref(out), ctypes.byref(n))
        return ctypes.string_at(out, n.value).decode()

    # ---- hot path -------------------------------------------------
    def raw_search(self, queries, k, nprobe=0, rerank=0, metric=0,
                   request_id=""):
        q = np.ascontiguousarray(queries, dtype=np.float32)
        nq = q.shape[0]
        dists = np.empty((nq, k), dtype=np.float32)
        ids = np.empty((nq, k), dtype=np.int64)
        rc = lib().GammaRawSearch(self.h, nq, _fp(q), k, nprobe, rerank,
                                  metric, _fp(dists), _ip(ids))
        if rc == -2:
            raise InterruptedError("search killed")
        if rc != 0:
            raise RuntimeError(f"GammaRawSearch failed rc={rc}")
        return dists, ids

    def cache_queries(self, queries):
        q = np.ascontiguousarray(queries, dtype=np.float32)
        rc = lib().GammaCacheQueries(self.h, q.shape[0], _fp(q))
        if rc != 0:
            raise RuntimeError("GammaCacheQueries failed")
        return q.shape[0]

    def search_cached(self, nq, k, nprobe=0, rerank=0, metric=0):
        dists = np.empty((nq, k), dtype=np.float32)
        ids = np.empty((nq, k), dtype=np.int64)
        rc = lib().GammaRawSearchCached(self.h, nq, k, nprobe, rerank,
                                        metric, _fp(dists), _ip(ids))
        if rc == -2:
            raise InterruptedError("search killed")
        if rc != 0:
            raise RuntimeError(f"GammaRawSearchCached failed rc={rc}")
        return dists, ids

    def search_pb(self, queries, topn, index_params="", fields=("_id",),
                  request_id="req1", partition_id=1, brute=0,
                  min_score=None, max_score=None, l2_sqrt=False,
                  term_filters=(), range_filters=(), operator=0,
                  extra_vec_queries=(), multi_vector_rank=0, ranker="",
                  offset=0):
        """The real C-ABI Search with protobuf marshalling (reader.go
        path). extra_vec_queries: [(field_name, query_array)] for
        multi-vector search; multi_vector_rank orders by combined
        score; ranker = WeightedRanker JSON."""
        from . import proto
        q = np.ascontiguousarray(queries, dtype=np.float32)
        extra = []
        for ef in extra_vec_queries:
            row = [ef[0],
                   np.ascontiguousarray(ef[1], np.float32).tobytes()]
            row += list(ef[2:])  # optional (min_score, max_score)
            extra.append(tuple(row))
        req = proto.encode_search_request(
            self.vec_name, q.tobytes(), topn, q.shape[0],
            request_id=request_id, partition_id=partition_id,
            index_params=index_params, brute=brute, fields=fields,
            min_score=min_score, max_score=max_score, l2_sqrt=l2_sqrt,
            term_filters=term_filters, range_filters=range_filters,
            operator=operator, extra_vec_fields=extra,
            multi_vector_rank=multi_vector_rank, ranker=ranker,
            offset=offset)
        out = ctypes.c_char_p()
        n = ctypes.c_int()
        st = lib().Search(self.h, req, len(req), ctypes.byref(out),
                          ctypes.byref(n))
        if st.code == -2:
            raise InterruptedError("search killed")
        _check(st, "Search")
        buf = ctypes.string_at(out, n.value)
        return proto.decode_search_response(buf)

    def query_pb(self, document_ids=(), fields=("_id",), term_filters=(),
                 range_filters=(), limit=0, operator=0,
                 is_vector_value=False):
        """The C-ABI Query (doc fetch / filtered browse)."""
        from . import proto
        req = proto.encode_query_request(
            list(document_ids), fields=fields, term_filters=term_filters,
            range_filters=range_filters, limit=limit, operator=operator,
            is_vector_value=is_vector_value)
        out = ctypes.c_char_p()
        n = ctypes.c_int()
        st = lib().Query(self.h, req, len(req), ctypes.byref(out),
                         ctypes.byref(n))
        _check(st, "Query")
        buf = ctypes.string_at(out, n.value)
        return proto.decode_search_response(buf)

    def last_timing(self):
        t = (ctypes.c_double * 6)()
        lib().GammaLastSearchTiming(self.h, t)
        return {k: t[i] for i, k in enumerate(
            ["h2d_us", "assign_us", "scan_us", "post_us", "d2h_us",
             "total_us"])}

    # ---- debug hooks for parity tests ------------------------------
    def debug_coarse_assign(self, queries, nprobe):
        q = np.ascontiguousarray(queries, dtype=np.float32)
        nq = q.shape[0]
        lists = np.empty((nq, nprobe), dtype=np.int64)
        dists = np.empty((nq, nprobe), dtype=np.float32)
        rc = lib().GammaDebugCoarseAssign(self.h, nq, _fp(q), nprobe,
                                          _ip(lists), _fp(dists))
        if rc != 0:
            raise RuntimeError("debug_coarse_assign failed")
        return dists, lists

    def debug_model(self, nlist, d, M=0, ksub=256):
        cent = np.empty((nlist, d), dtype=np.float32)
        books = (np.empty((M, ksub, d // M), dtype=np.float32)
                 if M else np.empty(0, dtype=np.float32))
        rc = lib().GammaDebugGetModel(self.h, _fp(cent),
                                      _fp(books) if M else None)
        if rc != 0:
            raise RuntimeError("debug_model failed")
        return cent, books

    def debug_opq(self, d):
        """d x d OPQ rotation R (row-major, y = R x); raises if none."""
        R = np.zeros((d, d), dtype=np.float32)
        if lib().GammaDebugGetOPQ(self.h, _fp(R)) != 0:
            raise RuntimeError("no OPQ")
        return R

    def debug_apply_opq(self, xq):
        """engine-GPU-rotated queries (bit-exact input for the oracle)."""
        xq = np.ascontiguousarray(xq, dtype=np.float32)
        out = np.zeros_like(xq)
        if lib().GammaDebugApplyOPQ(self.h, _fp(xq), xq.shape[0],
                                    _fp(out)) != 0:
            raise RuntimeError("no OPQ")
        return out

    def debug_list(self, list_no, code_size):
        n = lib().GammaDebugGetList(self.h, list_no, None, None)
        if n < 0:
            raise RuntimeError("debug_list failed")
        if n == 0:
            return (np.empty(0, dtype=np.int64),
                    np.empty((0, code_size), dtype=np.uint8))
        ids = np.empty(n, dtype=np.int64)
        codes = np.empty((n, code_size), dtype=np.uint8)
        lib().GammaDebugGetList(
            self.h, list_no, _ip(ids),
            codes.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
        return ids, codes


def set_kill(request_id, partition_id, reason=1):
    lib().SetKillStatus(request_id.encode(), partition_id, reason)


def clear_kill(request_id, partition_id):
    lib().DeleteKillStatus(request_id.encode(), partition_id)
