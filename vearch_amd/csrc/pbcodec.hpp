/*
 * pbcodec.hpp — hand-written protobuf wire-format codec (header-only) for
 * the message subset the Gamma C ABI exchanges:
 *   decode vearchpb.SearchRequest / QueryRequest
 *     (internal/proto/router_grpc.proto:146-192)
 *   encode vearchpb.SearchResponse
 *     (router_grpc.proto:193-221, built as c_api/api_data/response.cc does)
 * Written from the .proto field numbers; no protobuf library involved.
 */
#pragma once
#include <stdint.h>
#include <string.h>

#include <map>
#include <string>
#include <vector>

namespace gpb {

/* ------------------------------------------------------------- low level */
struct Reader {
  const uint8_t *p, *end;
  Reader(const void *buf, size_t n)
      : p((const uint8_t *)buf), end((const uint8_t *)buf + n) {}
  bool ok() const { return p <= end; }
  bool done() const { return p >= end; }
  bool varint(uint64_t &v) {
    v = 0;
    int shift = 0;
    while (p < end && shift < 64) {
      uint8_t b = *p++;
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return true;
      shift += 7;
    }
    return false;
  }
  bool key(uint32_t &field, uint32_t &wt) {
    uint64_t k;
    if (done() || !varint(k)) return false;
    field = (uint32_t)(k >> 3);
    wt = (uint32_t)(k & 7);
    return true;
  }
  bool bytes(std::string &out) {
    uint64_t n;
    if (!varint(n) || (uint64_t)(end - p) < n) return false;
    out.assign((const char *)p, n);
    p += n;
    return true;
  }
  bool sub(Reader &r) {
    uint64_t n;
    if (!varint(n) || (uint64_t)(end - p) < n) return false;
    r = Reader(p, n);
    p += n;
    return true;
  }
  bool fixed64(uint64_t &v) {
    if (end - p < 8) return false;
    memcpy(&v, p, 8);
    p += 8;
    return true;
  }
  bool fixed32(uint32_t &v) {
    if (end - p < 4) return false;
    memcpy(&v, p, 4);
    p += 4;
    return true;
  }
  bool skip(uint32_t wt) {
    uint64_t tmp;
    switch (wt) {
      case 0: return varint(tmp);
      case 1: return fixed64(tmp);
      case 2: {
        uint64_t n;
        if (!varint(n) || (uint64_t)(end - p) < n) return false;
        p += n;
        return true;
      }
      case 5: {
        uint32_t t;
        return fixed32(t);
      }
      default: return false;
    }
  }
};

struct Writer {
  std::string out;
  void raw_varint(uint64_t v) {
    while (v >= 0x80) {
      out.push_back((char)(v | 0x80));
      v >>= 7;
    }
    out.push_back((char)v);
  }
  void tag(uint32_t field, uint32_t wt) { raw_varint((field << 3) | wt); }
  void v_int(uint32_t field, int64_t v) {
    if (v == 0) return;
    tag(field, 0);
    raw_varint((uint64_t)v);
  }
  void v_bool(uint32_t field, bool b) { v_int(field, b ? 1 : 0); }
  void v_double(uint32_t field, double d) {
    tag(field, 1);
    uint64_t u;
    memcpy(&u, &d, 8);
    for (int i = 0; i < 8; i++) out.push_back((char)((u >> (8 * i)) & 0xff));
  }
  void v_str(uint32_t field, const std::string &s, bool always = false) {
    if (s.empty() && !always) return;
    tag(field, 2);
    raw_varint(s.size());
    out.append(s);
  }
  void v_msg(uint32_t field, const std::string &sub) {
    tag(field, 2);
    raw_varint(sub.size());
    out.append(sub);
  }
};

/* ------------------------------------------------------- SearchRequest */
struct VectorQuery { /* router_grpc.proto:128-135 */
  std::string name;
  std::string value; /* raw fp32 bytes; n = len/(4d), vector_manager.cc:915 */
  double min_score = -3.402823466e38;
  double max_score = 3.402823466e38;
  bool has_min = false, has_max = false;
  std::string format, index_type;
};

struct TermFilter { /* router_grpc.proto:110-114 */
  std::string field;
  std::string value; /* one or more terms separated by \x01 */
  int is_union = 0;
};
struct RangeFilter { /* router_grpc.proto:116-123; values = raw binary of
                        the field type (NumericToSortableStr convention,
                        table/inverted_index.cc:263) */
  std::string field, lower, upper;
  bool include_lower = false, include_upper = false;
  int is_union = 0; /* FilterOperator: And=0 Or=1 Not=2 (engine.cc:475) */
};

struct SearchRequest { /* router_grpc.proto:168-192 */
  std::string request_id;
  int partition_id = 0;
  int req_num = 0, topn = 0, brute = 0;
  std::vector<VectorQuery> vec_fields;
  std::vector<std::string> fields;
  std::vector<RangeFilter> range_filters;
  std::vector<TermFilter> term_filters;
  int n_filters = 0;
  std::string index_params, ranker;
  int multi_vector_rank = 0;
  bool l2_sqrt = false, trace = false, is_vector_value = false;
  int op = 0, offset = 0;

  bool parse(const char *buf, int len) {
    Reader r(buf, (size_t)len);
    uint32_t f, wt;
    while (!r.done()) {
      if (!r.key(f, wt)) return false;
      switch (f) {
        case 1: { /* head.params: request_id, partition_id (request.cc:25-34) */
          Reader h(nullptr, 0);
          if (!r.sub(h)) return false;
          uint32_t hf, hwt;
          while (!h.done()) {
            if (!h.key(hf, hwt)) return false;
            if (hf == 7 && hwt == 2) { /* params map entry */
              Reader e(nullptr, 0);
              if (!h.sub(e)) return false;
              std::string key, val;
              uint32_t ef, ewt;
              while (!e.done()) {
                if (!e.key(ef, ewt)) return false;
                if (ef == 1 && ewt == 2) {
                  if (!e.bytes(key)) return false;
                } else if (ef == 2 && ewt == 2) {
                  if (!e.bytes(val)) return false;
                } else if (!e.skip(ewt)) {
                  return false;
                }
              }
              if (key == "request_id") request_id = val;
              else if (key == "partition_id") partition_id = atoi(val.c_str());
            } else if (!h.skip(hwt)) {
              return false;
            }
          }
          break;
        }
        case 2: { uint64_t v; if (!r.varint(v)) return false; req_num = (int)v; break; }
        case 3: { uint64_t v; if (!r.varint(v)) return false; topn = (int)v; break; }
        case 4: { uint64_t v; if (!r.varint(v)) return false; brute = (int)v; break; }
        case 5: { /* vec_fields */
          Reader q(nullptr, 0);
          if (!r.sub(q)) return false;
          VectorQuery vq;
          uint32_t qf, qwt;
          while (!q.done()) {
            if (!q.key(qf, qwt)) return false;
            switch (qf) {
              case 1: if (!q.bytes(vq.name)) return false; break;
              case 2: if (!q.bytes(vq.value)) return false; break;
              case 3: { uint64_t u; if (!q.fixed64(u)) return false;
                        memcpy(&vq.min_score, &u, 8); vq.has_min = true; break; }
              case 4: { uint64_t u; if (!q.fixed64(u)) return false;
                        memcpy(&vq.max_score, &u, 8); vq.has_max = true; break; }
              case 5: if (!q.bytes(vq.format)) return false; break;
              case 6: if (!q.bytes(vq.index_type)) return false; break;
              default: if (!q.skip(qwt)) return false;
            }
          }
          vec_fields.push_back(std::move(vq));
          break;
        }
        case 6: { std::string s; if (!r.bytes(s)) return false;
                  fields.push_back(std::move(s)); break; }
        case 7: { /* range_filters */
          Reader fr(nullptr, 0);
          if (!r.sub(fr)) return false;
          RangeFilter rf;
          uint32_t ff, fwt;
          while (!fr.done()) {
            if (!fr.key(ff, fwt)) return false;
            switch (ff) {
              case 1: if (!fr.bytes(rf.field)) return false; break;
              case 2: if (!fr.bytes(rf.lower)) return false; break;
              case 3: if (!fr.bytes(rf.upper)) return false; break;
              case 4: { uint64_t v; if (!fr.varint(v)) return false;
                        rf.include_lower = v != 0; break; }
              case 5: { uint64_t v; if (!fr.varint(v)) return false;
                        rf.include_upper = v != 0; break; }
              case 6: { uint64_t v; if (!fr.varint(v)) return false;
                        rf.is_union = (int)v; break; }
              default: if (!fr.skip(fwt)) return false;
            }
          }
          range_filters.push_back(std::move(rf));
          n_filters++;
          break;
        }
        case 8: { /* term_filters */
          Reader fr(nullptr, 0);
          if (!r.sub(fr)) return false;
          TermFilter tf;
          uint32_t ff, fwt;
          while (!fr.done()) {
            if (!fr.key(ff, fwt)) return false;
            switch (ff) {
              case 1: if (!fr.bytes(tf.field)) return false; break;
              case 2: if (!fr.bytes(tf.value)) return false; break;
              case 3: { uint64_t v; if (!fr.varint(v)) return false;
                        tf.is_union = (int)v; break; }
              default: if (!fr.skip(fwt)) return false;
            }
          }
          term_filters.push_back(std::move(tf));
          n_filters++;
          break;
        }
        case 9: if (!r.bytes(index_params)) return false; break;
        case 10: { uint64_t v; if (!r.varint(v)) return false;
                   multi_vector_rank = (int)v; break; }
        case 11: { uint64_t v; if (!r.varint(v)) return false;
                   l2_sqrt = v != 0; break; }
        case 15: if (!r.bytes(ranker)) return false; break;
        case 16: { uint64_t v; if (!r.varint(v)) return false;
                   trace = v != 0; break; }
        case 17: { uint64_t v; if (!r.varint(v)) return false; op = (int)v; break; }
        case 20: { uint64_t v; if (!r.varint(v)) return false;
                   offset = (int)v; break; }
        case 12: { uint64_t v; if (!r.varint(v)) return false;
                   is_vector_value = v != 0; break; }
        default: if (!r.skip(wt)) return false;
      }
    }
    return true;
  }
};

/* ------------------------------------------------------- QueryRequest */
struct QueryRequest { /* router_grpc.proto:146-166 */
  std::string request_id;
  int partition_id = 0;
  std::vector<std::string> document_ids;
  std::vector<std::string> fields;
  std::vector<RangeFilter> range_filters;
  std::vector<TermFilter> term_filters;
  bool is_vector_value = false;
  int limit = 0;
  int n_filters = 0;
  int op = 0; /* request-level FilterOperator (`operator` field 15) */

  bool parse(const char *buf, int len) {
    Reader r(buf, (size_t)len);
    uint32_t f, wt;
    while (!r.done()) {
      if (!r.key(f, wt)) return false;
      switch (f) {
        case 1: { /* head */
          Reader h(nullptr, 0);
          if (!r.sub(h)) return false;
          uint32_t hf, hwt;
          while (!h.done()) {
            if (!h.key(hf, hwt)) return false;
            if (hf == 7 && hwt == 2) {
              Reader e(nullptr, 0);
              if (!h.sub(e)) return false;
              std::string key, val;
              uint32_t ef, ewt;
              while (!e.done()) {
                if (!e.key(ef, ewt)) return false;
                if (ef == 1 && ewt == 2) { if (!e.bytes(key)) return false; }
                else if (ef == 2 && ewt == 2) { if (!e.bytes(val)) return false; }
                else if (!e.skip(ewt)) return false;
              }
              if (key == "request_id") request_id = val;
              else if (key == "partition_id") partition_id = atoi(val.c_str());
            } else if (!h.skip(hwt)) return false;
          }
          break;
        }
        case 2: { std::string s; if (!r.bytes(s)) return false;
                  document_ids.push_back(std::move(s)); break; }
        case 5: { /* range_filters */
          Reader fr(nullptr, 0);
          if (!r.sub(fr)) return false;
          RangeFilter rf;
          uint32_t ff, fwt;
          while (!fr.done()) {
            if (!fr.key(ff, fwt)) return false;
            switch (ff) {
              case 1: if (!fr.bytes(rf.field)) return false; break;
              case 2: if (!fr.bytes(rf.lower)) return false; break;
              case 3: if (!fr.bytes(rf.upper)) return false; break;
              case 4: { uint64_t v; if (!fr.varint(v)) return false;
                        rf.include_lower = v != 0; break; }
              case 5: { uint64_t v; if (!fr.varint(v)) return false;
                        rf.include_upper = v != 0; break; }
              case 6: { uint64_t v; if (!fr.varint(v)) return false;
                        rf.is_union = (int)v; break; }
              default: if (!fr.skip(fwt)) return false;
            }
          }
          range_filters.push_back(std::move(rf));
          n_filters++;
          break;
        }
        case 6: { /* term_filters */
          Reader fr(nullptr, 0);
          if (!r.sub(fr)) return false;
          TermFilter tf;
          uint32_t ff, fwt;
          while (!fr.done()) {
            if (!fr.key(ff, fwt)) return false;
            switch (ff) {
              case 1: if (!fr.bytes(tf.field)) return false; break;
              case 2: if (!fr.bytes(tf.value)) return false; break;
              case 3: { uint64_t v; if (!fr.varint(v)) return false;
                        tf.is_union = (int)v; break; }
              default: if (!fr.skip(fwt)) return false;
            }
          }
          term_filters.push_back(std::move(tf));
          n_filters++;
          break;
        }
        case 7: { std::string s; if (!r.bytes(s)) return false;
                  fields.push_back(std::move(s)); break; }
        case 8: { uint64_t v; if (!r.varint(v)) return false;
                  is_vector_value = v != 0; break; }
        case 9: { uint64_t v; if (!r.varint(v)) return false;
                  limit = (int)v; break; }
        case 15: { uint64_t v; if (!r.varint(v)) return false;
                   op = (int)v; break; }
        default: if (!r.skip(wt)) return false;
      }
    }
    return true;
  }
};

/* ------------------------------------------------------ SearchResponse
 * Built the way response.cc builds vearchpb.SearchResponse:
 *   results(2): per query SearchResult {
 *     max_score(2), status(5){total(1),failed(2),successful(3),msg(4)},
 *     msg(6), result_items(7){score(1), fields(2){name(1), value(3)},
 *     p_key(3)}, timeout(9) }
 */
struct ResultField {
  std::string name;
  std::string value;
};
struct ResultItem {
  double score = 0;
  std::vector<ResultField> fields;
};
struct SearchResult {
  int total = 0;
  double max_score = -1.7976931348623157e308;
  std::string msg;
  std::vector<ResultItem> items;
};

inline std::string encode_search_response(
    const std::vector<SearchResult> &results) {
  Writer w;
  for (const auto &res : results) {
    Writer r;
    r.v_double(2, res.max_score);
    {
      Writer st;
      st.v_int(1, res.total);
      st.v_int(2, 0);
      st.v_int(3, res.total);
      st.v_str(4, res.msg.empty() ? std::string("success") : res.msg);
      r.v_msg(5, st.out);
    }
    r.v_str(6, res.msg);
    for (const auto &it : res.items) {
      Writer item;
      item.v_double(1, it.score);
      for (const auto &fv : it.fields) {
        Writer fw;
        fw.v_str(1, fv.name);
        fw.v_str(3, fv.value, true);
        item.v_msg(2, fw.out);
      }
      r.v_msg(7, item.out);
    }
    w.v_msg(2, r.out);
  }
  return w.out;
}

}  // namespace gpb
