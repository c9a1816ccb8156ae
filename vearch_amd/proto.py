"""Protobuf wire-format encoder/decoder for the vearchpb messages the test
and bench harness exchanges with libgamma.so, playing the role of the Go
caller (internal/proto/router_grpc.proto). No protoc involved."""
import struct


def _varint(v):
    out = bytearray()
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    return bytes(out)


def _tag(field, wt):
    return _varint((field << 3) | wt)


def _ld(field, data):
    return _tag(field, 2) + _varint(len(data)) + data


def _vint(field, v):
    return b"" if v == 0 else _tag(field, 0) + _varint(v)


def _vdouble(field, v):
    return _tag(field, 1) + struct.pack("<d", v)


def encode_head(request_id="", partition_id=None):
    params = b""
    if request_id:
        params += _ld(7, _ld(1, b"request_id") + _ld(2, request_id.encode()))
    if partition_id is not None:
        params += _ld(
            7, _ld(1, b"partition_id") + _ld(2, str(partition_id).encode()))
    return params


def encode_search_request(vec_name, queries_f32_bytes, topn, req_num,
                          request_id="req1", partition_id=1,
                          index_params="", min_score=None, max_score=None,
                          brute=0, fields=("_id",), l2_sqrt=False,
                          is_vector_value=False, term_filters=(),
                          range_filters=(), operator=0,
                          extra_vec_fields=(), multi_vector_rank=0,
                          ranker="", offset=0):
    """vearchpb.SearchRequest (router_grpc.proto:168-192).
    term_filters: (field, value_bytes) or (field, value_bytes, is_union).
    range_filters: (field, lower_bytes, upper_bytes, incl_lower,
    incl_upper[, is_union]) — values are the raw binary of the field
    type. is_union is the per-filter FilterOperator (And=0 Or=1 Not=2,
    engine.cc:475); `operator` is the request-level combiner (And=0
    intersects the filters' match sets, Or=1 unions them,
    scalar_index_manager.cc:1188)."""
    out = b""
    out += _ld(1, encode_head(request_id, partition_id))
    out += _vint(2, req_num)
    out += _vint(3, topn)
    out += _vint(4, brute)
    vq = _ld(1, vec_name.encode()) + _ld(2, queries_f32_bytes)
    if min_score is not None:
        vq += _vdouble(3, min_score)
    if max_score is not None:
        vq += _vdouble(4, max_score)
    out += _ld(5, vq)
    for ef in extra_vec_fields:
        evq = _ld(1, ef[0].encode()) + _ld(2, ef[1])
        if len(ef) > 2 and ef[2] is not None:
            evq += _vdouble(3, ef[2])
        if len(ef) > 3 and ef[3] is not None:
            evq += _vdouble(4, ef[3])
        out += _ld(5, evq)
    for f in fields:
        out += _ld(6, f.encode())
    for rf in range_filters:
        field, lower, upper, il, iu = rf[:5]
        body = _ld(1, field.encode()) + _ld(2, lower) + _ld(3, upper)
        body += _vint(4, 1 if il else 0) + _vint(5, 1 if iu else 0)
        if len(rf) > 5:
            body += _vint(6, rf[5])
        out += _ld(7, body)
    for tf in term_filters:
        field, value = tf[0], tf[1]
        body = _ld(1, field.encode()) + _ld(2, value)
        if len(tf) > 2:
            body += _vint(3, tf[2])
        out += _ld(8, body)
    if index_params:
        out += _ld(9, index_params.encode())
    if multi_vector_rank:
        out += _vint(10, multi_vector_rank)
    if l2_sqrt:
        out += _vint(11, 1)
    if is_vector_value:
        out += _vint(12, 1)
    if ranker:
        out += _ld(15, ranker.encode())
    if operator:
        out += _vint(17, operator)
    if offset:
        out += _vint(20, offset)
    return out


def encode_query_request(document_ids, fields=("_id",), request_id="req1",
                         partition_id=1, is_vector_value=False,
                         term_filters=(), range_filters=(), limit=0,
                         operator=0):
    out = b""
    out += _ld(1, encode_head(request_id, partition_id))
    for d in document_ids:
        out += _ld(2, d.encode())
    for rf in range_filters:
        field, lower, upper, il, iu = rf[:5]
        body = _ld(1, field.encode()) + _ld(2, lower) + _ld(3, upper)
        body += _vint(4, 1 if il else 0) + _vint(5, 1 if iu else 0)
        if len(rf) > 5:
            body += _vint(6, rf[5])
        out += _ld(5, body)
    for tf in term_filters:
        body = _ld(1, tf[0].encode()) + _ld(2, tf[1])
        if len(tf) > 2:
            body += _vint(3, tf[2])
        out += _ld(6, body)
    for f in fields:
        out += _ld(7, f.encode())
    if is_vector_value:
        out += _vint(8, 1)
    if limit:
        out += _vint(9, limit)
    if operator:
        out += _vint(15, operator)
    return out


# ------------------------------------------------------------- decoding
def _read_varint(buf, i):
    v = 0
    shift = 0
    while True:
        b = buf[i]
        i += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, i
        shift += 7


def _fields(buf):
    i = 0
    n = len(buf)
    while i < n:
        key, i = _read_varint(buf, i)
        f, wt = key >> 3, key & 7
        if wt == 0:
            v, i = _read_varint(buf, i)
        elif wt == 1:
            v = struct.unpack("<d", buf[i:i + 8])[0]
            i += 8
        elif wt == 2:
            ln, i = _read_varint(buf, i)
            v = buf[i:i + ln]
            i += ln
        elif wt == 5:
            v = struct.unpack("<f", buf[i:i + 4])[0]
            i += 4
        else:
            raise ValueError(f"wiretype {wt}")
        yield f, wt, v


def decode_search_response(buf):
    """-> list per query: {'max_score','total','items':[{'score','fields':
    {name: bytes}}]} (router_grpc.proto:195-216 SearchResult subset)."""
    results = []
    for f, wt, v in _fields(buf):
        if f != 2:
            continue
        res = {"max_score": None, "total": None, "msg": "", "items": []}
        for rf, rwt, rv in _fields(v):
            if rf == 2:
                res["max_score"] = rv
            elif rf == 5:
                for sf, _, sv in _fields(rv):
                    if sf == 1:
                        res["total"] = sv
            elif rf == 6:
                res["msg"] = rv.decode(errors="replace")
            elif rf == 7:
                item = {"score": None, "fields": {}}
                for itf, _, itv in _fields(rv):
                    if itf == 1:
                        item["score"] = itv
                    elif itf == 2:
                        fd = {"name": None, "value": b""}
                        for ff, _, fvv in _fields(itv):
                            if ff == 1:
                                fd["name"] = fvv.decode()
                            elif ff == 3:
                                fd["value"] = fvv
                        item["fields"][fd["name"]] = fd["value"]
                res["items"].append(item)
        results.append(res)
    return results
