"""CPU-side tests of the C ABI library: it loads, exports exactly the
reference's 20+ entry points (gamma_api.h:26-198), and the hand-written
protobuf / FlatBuffers codecs agree with the harness encoders. No compute
here (no GPU in CI)."""
import ctypes
import json
import os
import subprocess

import numpy as np
import pytest

import __graft_entry__ as entry
from vearch_amd import fbsenc, proto

REQUIRED_EXPORTS = [
    # include/gamma_api.h (reference gamma_api.h:26-198)
    "Init", "Close", "CreateTable", "AddOrUpdateDoc", "DeleteDoc",
    "GetEngineStatus", "GetMemoryInfo", "GetDocByID", "GetDocByDocID",
    "BuildIndex", "RebuildIndex", "Dump", "Load", "Search", "Query",
    "SetConfig", "GetConfig", "Backup", "AddFieldIndexWithParams",
    "RemoveFieldIndex", "SetMemoryLimitConfig", "SetKillStatus",
    "DeleteKillStatus",
]


@pytest.fixture(scope="module")
def libpath():
    entry.build()
    return entry.SO


@pytest.fixture(scope="module")
def L(libpath):
    return ctypes.CDLL(libpath)


def test_exports(libpath):
    out = subprocess.check_output(["nm", "-D", libpath]).decode()
    syms = {line.split()[-1] for line in out.splitlines()
            if " T " in line}
    missing = [s for s in REQUIRED_EXPORTS if s not in syms]
    assert not missing, f"missing C-ABI exports: {missing}"


def _parse(L, fn, buf):
    out = ctypes.c_char_p()
    n = ctypes.c_int()
    rc = fn(buf, len(buf), ctypes.byref(out), ctypes.byref(n))
    assert rc == 0
    return json.loads(ctypes.string_at(out, n.value).decode())


def test_search_request_codec(L):
    q = np.arange(256, dtype=np.float32)
    buf = proto.encode_search_request(
        "emb", q.tobytes(), topn=10, req_num=2, request_id="rid42",
        partition_id=7, index_params='{"nprobe": 32}', min_score=-1e9,
        max_score=1e9, brute=1, fields=("_id", "tag"), l2_sqrt=True)
    L.GammaTestParseSearchRequest.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.POINTER(ctypes.c_char_p),
        ctypes.POINTER(ctypes.c_int)]
    j = _parse(L, L.GammaTestParseSearchRequest, buf)
    assert j["request_id"] == "rid42"
    assert j["partition_id"] == 7
    assert j["req_num"] == 2 and j["topn"] == 10 and j["brute"] == 1
    assert j["n_vec"] == 1 and j["vec_name"] == "emb"
    assert j["vec_bytes"] == 1024
    assert j["index_params"] == '{"nprobe": 32}'
    assert j["l2_sqrt"] == 1
    assert j["n_fields"] == 2 and j["n_filters"] == 0
    assert j["op"] == 0 and j["term_unions"] == [] and j["range_unions"] == []


def test_search_request_codec_filter_operators(L):
    """is_union on term/range filters (FilterOperator And=0 Or=1 Not=2,
    engine.cc:475) and the request-level `operator` (field 17) survive
    the wire round trip."""
    q = np.arange(128, dtype=np.float32)
    buf = proto.encode_search_request(
        "emb", q.tobytes(), topn=5, req_num=1,
        term_filters=[("tag", b"a", 2), ("tag2", b"b")],
        range_filters=[("num", b"\x00" * 4, b"\x00" * 4, True, True, 2),
                       ("num2", b"\x00" * 4, b"\x01" * 4, False, True)],
        operator=1)
    L.GammaTestParseSearchRequest.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.POINTER(ctypes.c_char_p),
        ctypes.POINTER(ctypes.c_int)]
    j = _parse(L, L.GammaTestParseSearchRequest, buf)
    assert j["op"] == 1
    assert j["term_unions"] == [2, 0]
    assert j["range_unions"] == [2, 0]
    assert j["n_filters"] == 4


def test_table_codec(L):
    buf = fbsenc.build_table(
        "space1", [("tag", fbsenc.DATA_STRING), ("num", fbsenc.DATA_INT)],
        "vecf", 128, "IVFPQ",
        '{"ncentroids": 256, "nsubvector": 32}')
    L.GammaTestParseTable.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.POINTER(ctypes.c_char_p),
        ctypes.POINTER(ctypes.c_int)]
    j = _parse(L, L.GammaTestParseTable, buf)
    assert j["name"] == "space1"
    assert j["index_type"] == "IVFPQ"
    assert j["n_fields"] == 2 and j["n_vectors"] == 1
    assert j["vec_name"] == "vecf" and j["dimension"] == 128
    assert j["field_names"] == "tag,num,"
    assert "ncentroids" in j["index_params"]


def test_doc_codec_roundtrip(L):
    vec = np.arange(32, dtype=np.float32).tobytes()
    buf = fbsenc.build_doc([
        ("_id", b"doc7", fbsenc.DATA_STRING),
        ("tag", b"hello", fbsenc.DATA_STRING),
        ("emb", vec, fbsenc.DATA_VECTOR),
    ])
    L.GammaTestDocRoundtrip.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.POINTER(ctypes.c_char_p),
        ctypes.POINTER(ctypes.c_int)]
    out = ctypes.c_char_p()
    n = ctypes.c_int()
    rc = L.GammaTestDocRoundtrip(buf, len(buf), ctypes.byref(out),
                                 ctypes.byref(n))
    assert rc == 0, f"doc roundtrip failed rc={rc}"


def test_search_request_codec_randomized(L):
    """Property-style: 60 random valid SearchRequests round-trip every
    header field through the C parser (arbitrary names/sizes/filters —
    the Go router composes requests with all of these)."""
    import random
    rng = random.Random(99)
    L.GammaTestParseSearchRequest.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.POINTER(ctypes.c_char_p),
        ctypes.POINTER(ctypes.c_int)]
    for trial in range(60):
        d = rng.choice([1, 4, 63, 128, 770])
        nvec = 1
        name = "".join(rng.choice("abz_09") for _ in range(
            rng.randint(1, 24)))
        rid = "".join(rng.choice("r0-9") for _ in range(rng.randint(0, 12)))
        topn = rng.randint(1, 1000)
        req_num = rng.randint(1, 64)
        pid = rng.randint(0, 1 << 20)
        nterm = rng.randint(0, 3)
        nrange = rng.randint(0, 3)
        terms = [("f%d" % i,
                  bytes(rng.randrange(256) for _ in range(
                      rng.randint(0, 9))))
                 for i in range(nterm)]
        ranges = [("g%d" % i,
                   bytes(rng.randrange(256) for _ in range(4)),
                   bytes(rng.randrange(256) for _ in range(4)),
                   bool(rng.getrandbits(1)), bool(rng.getrandbits(1)))
                  for i in range(nrange)]
        q = np.arange(d * req_num, dtype=np.float32) + trial
        buf = proto.encode_search_request(
            name, q.tobytes(), topn=topn, req_num=req_num,
            request_id=rid, partition_id=pid,
            index_params='{"nprobe": %d}' % rng.randint(1, 512),
            term_filters=terms, range_filters=ranges)
        j = _parse(L, L.GammaTestParseSearchRequest, buf)
        assert j["request_id"] == rid
        assert j["partition_id"] == pid
        assert j["req_num"] == req_num and j["topn"] == topn
        assert j["vec_name"] == name
        assert j["vec_bytes"] == d * req_num * 4
        assert j["n_filters"] == nterm + nrange


def test_init_without_gpu_fails_loudly(L):
    """On a GPU-less box the engine must refuse to start (no silent CPU
    fallback)."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    L.Init.restype = ctypes.c_void_p
    L.Init.argtypes = [ctypes.c_char_p, ctypes.c_int]
    cfg = b'{"path": "/tmp/x"}'
    h = L.Init(cfg, len(cfg))
    assert not h


def test_kill_registry_roundtrip(L):
    L.SetKillStatus.argtypes = [ctypes.c_char_p, ctypes.c_int, ctypes.c_int]
    L.DeleteKillStatus.argtypes = [ctypes.c_char_p, ctypes.c_int]
    L.SetKillStatus(b"r1", 3, 1)
    L.DeleteKillStatus(b"r1", 3)


def test_search_response_decoder():
    """Python decoder against a hand-assembled wire message."""
    import struct

    def varint(v):
        out = b""
        while v >= 0x80:
            out += bytes([v & 0x7F | 0x80])
            v >>= 7
        return out + bytes([v])

    def ld(f, data):
        return varint((f << 3) | 2) + varint(len(data)) + data

    item = (bytes([0x09]) + struct.pack("<d", 1.5) +
            ld(2, ld(1, b"_id") + ld(3, b"k1")))
    res = (bytes([0x11]) + struct.pack("<d", 1.5) + ld(7, item))
    buf = ld(2, res)
    out = proto.decode_search_response(buf)
    assert len(out) == 1
    assert out[0]["max_score"] == 1.5
    assert out[0]["items"][0]["score"] == 1.5
    assert out[0]["items"][0]["fields"]["_id"] == b"k1"


def test_codec_fuzz_no_crash(L):
    """Mutated/truncated buffers must never crash the parsers — they
    either fail cleanly or produce bounded output (the Go side can feed
    arbitrary bytes across the C ABI)."""
    import random

    rng = random.Random(1234)
    q = np.arange(64, dtype=np.float32)
    pb = bytearray(proto.encode_search_request(
        "emb", q.tobytes(), topn=5, req_num=1,
        index_params='{"nprobe": 8}', term_filters=[("t", b"x")],
        range_filters=[("r", b"\x01\x00\x00\x00", b"\x05\x00\x00\x00",
                        True, True)]))
    fb = bytearray(fbsenc.build_table(
        "s", [("tag", fbsenc.DATA_STRING)], "v", 32, "IVFPQ", "{}"))
    doc = bytearray(fbsenc.build_doc(
        [("_id", b"k", fbsenc.DATA_STRING),
         ("v", np.arange(32, dtype=np.float32).tobytes(),
          fbsenc.DATA_VECTOR)]))
    L.GammaTestParseSearchRequest.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.POINTER(ctypes.c_char_p),
        ctypes.POINTER(ctypes.c_int)]
    L.GammaTestParseTable.argtypes = L.GammaTestParseSearchRequest.argtypes
    L.GammaTestDocRoundtrip.argtypes = L.GammaTestParseSearchRequest.argtypes

    def hammer(fn, base):
        for _ in range(400):
            b = bytearray(base)
            for _ in range(rng.randint(1, 6)):
                b[rng.randrange(len(b))] = rng.randrange(256)
            cut = rng.randint(0, len(b))
            buf = bytes(b[:cut]) if rng.random() < 0.3 else bytes(b)
            out = ctypes.c_char_p()
            n = ctypes.c_int()
            fn(buf, len(buf), ctypes.byref(out), ctypes.byref(n))

    hammer(L.GammaTestParseSearchRequest, pb)
    hammer(L.GammaTestParseTable, fb)
    hammer(L.GammaTestDocRoundtrip, doc)
