/*
 * capi.cpp — the drop-in C ABI (include/gamma_api.h) plus the bench/debug
 * extensions (include/gamma_bench.h). Mirrors the dispatch of the
 * reference's c_api/gamma_api.cc onto the MI355X engine.
 */
#include <math.h>
#include <stdlib.h>
#include <unistd.h>
#include <string.h>

#include <algorithm>
#include <string>
#include <vector>

#include "../../include/gamma_api.h"
#include "../../include/gamma_bench.h"
#include "core.hpp"
#include "fbs.hpp"
#include "json.hpp"
#include "pbcodec.hpp"

using vgamma::Engine;
using vgamma::FieldMeta;
using vgamma::KillRegistry;

static char *dup_malloc(const std::string &s) {
  char *p = (char *)malloc(s.size() + 1);
  memcpy(p, s.data(), s.size());
  p[s.size()] = 0;
  return p;
}

static CStatus ok_status() { return CStatus{0, nullptr}; }
static CStatus err_status(int code, const std::string &msg) {
  /* msg malloc'd (the Go side frees with C.free — gamma.go:186-188;
   * the reference's new[] at gamma_api.cc:150 was a latent mismatch) */
  return CStatus{code, dup_malloc(msg)};
}

static int g_memory_limit_mb = 0;

/* host RSS in MB (the reference's MemoryManager watermark check,
 * gamma_api.cc:180 / memory/memoryManager.cc) */
static long rss_mb() {
  FILE *f = fopen("/proc/self/statm", "r");
  if (!f) return 0;
  long pages = 0, rss = 0;
  if (fscanf(f, "%ld %ld", &pages, &rss) != 2) rss = 0;
  fclose(f);
  return rss * (sysconf(_SC_PAGESIZE) / 1024) / 1024;
}

static bool memory_exceeded() {
  return g_memory_limit_mb > 0 && rss_mb() > g_memory_limit_mb;
}

extern "C" {

static void *Init_unguarded(const char *config_str, int len) {
  auto *e = new Engine();
  std::string err;
  if (e->init(std::string(config_str ? config_str : "", len > 0 ? len : 0),
              &err)) {
    fprintf(stderr, "[gamma] Init failed: %s\n", err.c_str());
    delete e;
    return nullptr;
  }
  return e;
}

void *Init(const char *config_str, int len) {
  try {
    return Init_unguarded(config_str, len);
  } catch (...) {
    return nullptr;
  }
}


int Close(void *engine) {
  if (!engine) return 1;
  delete static_cast<Engine *>(engine);
  return 0;
}

static struct CStatus CreateTable_unguarded(void *engine, const char *table_str, int len) {
  if (!engine) return err_status(1, "null engine");
  gfb::TableSchema ts;
  if (!ts.parse(table_str, (size_t)len))
    return err_status(1, "bad table flatbuffer");
  if (ts.vectors.empty())
    return err_status(1, "table has no vector field");
  std::vector<FieldMeta> fields;
  for (auto &f : ts.fields) fields.push_back({f.name, f.data_type});
  /* extra vector fields of a multi-vector table (vector_manager.cc
   * keeps one index per field) */
  std::vector<std::pair<std::string, int>> extra_vecs;
  for (size_t i = 1; i < ts.vectors.size(); i++)
    extra_vecs.push_back({ts.vectors[i].name, ts.vectors[i].dimension});
  std::string index_type = ts.index_type;
  std::string index_params = ts.index_params;
  if (index_type.empty() && !ts.indexes.empty()) {
    index_type = ts.indexes[0].type;
    index_params = ts.indexes[0].params;
  }
  /* training_threshold comes from the index-params JSON (table.cc:122) */
  int tt = 0;
  {
    gjson::Value v;
    if (!index_params.empty() && gjson::parse(index_params, v))
      v.get_int("training_threshold", tt);
  }
  std::string err;
  if (static_cast<Engine *>(engine)->create_table(
          ts.name, fields, ts.vectors[0].name, ts.vectors[0].dimension,
          index_type, index_params, tt, &err, extra_vecs))
    return err_status(1, err);
  return ok_status();
}

struct CStatus CreateTable(void *engine, const char *table_str, int len) {
  try {
    return CreateTable_unguarded(engine, table_str, len);
  } catch (const std::exception &ex) {
    return err_status(-1,
                      std::string("internal error: ") + ex.what());
  } catch (...) {
    return err_status(-1, "internal error");
  }
}


static int AddOrUpdateDoc_unguarded(void *engine, const char *doc_str, int len) {
  if (!engine) return -1;
  if (memory_exceeded()) return -1;
  auto *e = static_cast<Engine *>(engine);
  gfb::Doc doc;
  if (!doc.parse(doc_str, (size_t)len)) return -1;
  std::string p_key;
  const float *vec = nullptr;
  int vec_len = 0;
  std::vector<std::pair<std::string, std::string>> fields;
  std::vector<vgamma::MultiVecQuery> extra;
  for (auto &f : doc.fields) {
    if (f.name == "_id") {
      p_key = f.value;
    } else if (f.data_type == gfb::VECTOR) {
      if (f.name == e->vec_field_name() ||
          (vec == nullptr && e->vec_dim_of(f.name) < 0)) {
        vec = (const float *)f.value.data();
        vec_len = (int)(f.value.size() / 4);
      } else {
        extra.push_back({f.name, (const float *)f.value.data()});
      }
    } else {
      fields.emplace_back(f.name, f.value);
    }
  }
  if (p_key.empty() || !vec) return -1;
  return e->add_doc(p_key, fields, vec, vec_len,
                    extra.empty() ? nullptr : &extra);
}

int AddOrUpdateDoc(void *engine, const char *doc_str, int len) {
  try {
    return AddOrUpdateDoc_unguarded(engine, doc_str, len);
  } catch (...) {
    return -1;
  }
}


static int DeleteDoc_unguarded(void *engine, const char *docid, int docid_len) {
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->delete_doc(
      std::string(docid, docid_len));
}

int DeleteDoc(void *engine, const char *docid, int docid_len) {
  try {
    return DeleteDoc_unguarded(engine, docid, docid_len);
  } catch (...) {
    return -1;
  }
}


void GetEngineStatus(void *engine, char **status, int *len) {
  std::string s = engine ? static_cast<Engine *>(engine)->status_json()
                         : std::string("{}");
  *status = dup_malloc(s);
  *len = (int)s.size();
}

void GetMemoryInfo(void *engine, char **memory_info, int *len) {
  char buf[256];
  auto *e = static_cast<Engine *>(engine);
  long long nd = e ? e->num_docs() : 0;
  long long dim = e ? e->dimension() : 0;
  snprintf(buf, sizeof buf,
           "{\"table_mem_bytes\": 0, \"vector_mem_bytes\": %lld, "
           "\"index_mem_bytes\": 0, \"bitmap_mem_bytes\": %lld}",
           nd * dim * 4, (nd + 7) / 8);
  std::string s(buf);
  *memory_info = dup_malloc(s);
  *len = (int)s.size();
}

static int serialize_doc(Engine *e, int64_t docid, char **doc_str,
                         int *len) {
  gfb::Doc doc;
  doc.fields.push_back({"_id", e->pkey_of(docid), gfb::STRING});
  for (auto &fm : e->scalar_fields()) {
    const std::string *v = e->field_value(docid, fm.name);
    doc.fields.push_back({fm.name, v ? *v : "", fm.data_type});
  }
  const float *vec = e->raw().host_row(docid);
  doc.fields.push_back(
      {e->vec_field_name(),
       std::string((const char *)vec, (size_t)e->dimension() * 4),
       gfb::VECTOR});
  for (auto &ev : e->extra_vec_fields()) {
    const float *v = ev->raw.host_row(docid);
    doc.fields.push_back(
        {ev->name, std::string((const char *)v, (size_t)ev->dim * 4),
         gfb::VECTOR});
  }
  std::string out = doc.serialize();
  *doc_str = dup_malloc(out);
  *len = (int)out.size();
  return 0;
}

static int GetDocByID_unguarded(void *engine, const char *docid, int docid_len,
               char **doc_str, int *len) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  auto rlock = e->read_lock(); /* stable doc state during serialization */
  int64_t id = e->docid_of(std::string(docid, docid_len));
  if (id < 0 || e->bitmap().test(id)) return -1;
  return serialize_doc(e, id, doc_str, len);
}

int GetDocByID(void *engine, const char *docid, int docid_len,
               char **doc_str, int *len) {
  try {
    return GetDocByID_unguarded(engine, docid, docid_len, doc_str, len);
  } catch (...) {
    return -1;
  }
}


static int GetDocByDocID_unguarded(void *engine, int docid, char next, char **doc_str,
                  int *len) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  auto rlock = e->read_lock(); /* stable doc state during serialization */
  int64_t id = docid;
  if (next) { /* next undeleted doc after docid (gamma_api.h:86) */
    id++;
    while (id < e->num_docs() && e->bitmap().test(id)) id++;
  }
  if (id < 0 || id >= e->num_docs() || e->bitmap().test(id)) return -1;
  return serialize_doc(e, id, doc_str, len);
}

int GetDocByDocID(void *engine, int docid, char next, char **doc_str,
                  int *len) {
  try {
    return GetDocByDocID_unguarded(engine, docid, next, doc_str, len);
  } catch (...) {
    return -1;
  }
}


static int BuildIndex_unguarded(void *engine) {
  if (!engine) return -1;
  std::string err;
  int rc = static_cast<Engine *>(engine)->build_index(&err);
  if (rc) fprintf(stderr, "[gamma] BuildIndex: %s\n", err.c_str());
  return rc;
}

int BuildIndex(void *engine) {
  try {
    return BuildIndex_unguarded(engine);
  } catch (...) {
    return -1;
  }
}


static int RebuildIndex_unguarded(void *engine, int drop_before_rebuild, int limit_cpu,
                 int describe) {
  (void)limit_cpu; /* GPU engine has no CPU throttle */
  (void)describe;
  if (!engine) return -1;
  std::string err;
  int rc = static_cast<Engine *>(engine)->rebuild_index(
      drop_before_rebuild != 0, &err);
  if (rc) fprintf(stderr, "[gamma] RebuildIndex: %s\n", err.c_str());
  return rc;
}

int RebuildIndex(void *engine, int drop_before_rebuild, int limit_cpu,
                 int describe) {
  try {
    return RebuildIndex_unguarded(engine, drop_before_rebuild, limit_cpu, describe);
  } catch (...) {
    return -1;
  }
}


static int Dump_unguarded(void *engine) {
  if (!engine) return -1;
  std::string err;
  int rc = static_cast<Engine *>(engine)->dump(&err);
  if (rc) fprintf(stderr, "[gamma] Dump: %s\n", err.c_str());
  return rc;
}

int Dump(void *engine) {
  try {
    return Dump_unguarded(engine);
  } catch (...) {
    return -1;
  }
}


static int Load_unguarded(void *engine) {
  if (!engine) return -1;
  std::string err;
  int rc = static_cast<Engine *>(engine)->load(&err);
  if (rc) fprintf(stderr, "[gamma] Load: %s\n", err.c_str());
  return rc;
}

int Load(void *engine) {
  try {
    return Load_unguarded(engine);
  } catch (...) {
    return -1;
  }
}


/* Multi-vector search (vector_manager.cc:851-1090): one query batch
 * over several vector fields, docid-intersection merge with
 * WeightedRanker scores. Request surface: vec_fields[] (one per
 * field), `ranker` = {"type":"WeightedRanker","params":[w...]}
 * (request.cc:86-92, default weights 1/vec_num),
 * `multi_vector_rank` != 0 -> results ordered by combined score
 * (:1073-1086), else docid-ascending. */
static struct CStatus SearchMulti_unguarded(Engine *e,
                                            const gpb::SearchRequest &req,
                                            char **response_str,
                                            int *res_len) {
  const size_t vn = req.vec_fields.size();
  int nprobe = 0, recall_num = 0, metric = 0;
  if (!req.index_params.empty()) {
    gjson::Value v;
    if (gjson::parse(req.index_params, v)) {
      v.get_int("nprobe", nprobe);
      v.get_int("recall_num", recall_num);
      std::string mt;
      if (v.get_str("metric_type", mt)) {
        if (strcasecmp(mt.c_str(), "L2") == 0) metric = 1;
        else if (strcasecmp(mt.c_str(), "InnerProduct") == 0 ||
                 strcasecmp(mt.c_str(), "IP") == 0)
          metric = 2;
        else
          return err_status(1, ("unknown metric_type: " + mt).c_str());
      }
    }
  }
  /* per-field queries; nq must agree across fields */
  std::vector<vgamma::MultiVecQuery> queries;
  int nq = -1;
  for (auto &vq : req.vec_fields) {
    int fd = e->vec_dim_of(vq.name);
    if (fd <= 0)
      return err_status(1, "unknown vector field " + vq.name);
    int fn = (int)(vq.value.size() / ((size_t)fd * 4));
    if (fn <= 0) return err_status(1, "Search n shouldn't less than 0!");
    if (nq < 0) nq = fn;
    else if (nq != fn)
      return err_status(1, "vector queries disagree on batch size");
    vgamma::MultiVecQuery mq;
    mq.name = vq.name;
    mq.vecs = (const float *)vq.value.data();
    mq.has_min = vq.has_min;
    mq.has_max = vq.has_max;
    mq.min_score = vq.min_score;
    mq.max_score = vq.max_score;
    queries.push_back(std::move(mq));
  }
  /* WeightedRanker (common_query_data.h:251-302) */
  std::vector<double> weights;
  if (!req.ranker.empty()) {
    gjson::Value rv;
    std::string msg = "weighted ranker params err: " + req.ranker;
    if (!gjson::parse(req.ranker, rv)) return err_status(1, msg);
    std::string rtype;
    if (!rv.get_str("type", rtype)) return err_status(1, msg);
    const gjson::Value *pv = rv.get("params");
    if (!pv || pv->type != gjson::Value::ARR)
      return err_status(1, msg);
    if (pv->arr.size() != vn)
      return err_status(1, "weighted ranker params: " + req.ranker +
                               ", length don't equal to " +
                               std::to_string(vn));
    for (auto &w : pv->arr) {
      if (w.type != gjson::Value::NUM) return err_status(1, msg);
      weights.push_back(w.num);
    }
  }
  std::vector<vgamma::TermFilterSpec> terms;
  for (auto &t : req.term_filters)
    terms.push_back({t.field, t.value, t.is_union});
  std::vector<vgamma::RangeFilterSpec> ranges;
  for (auto &t : req.range_filters)
    ranges.push_back({t.field, t.lower, t.upper, t.include_lower,
                      t.include_upper, t.is_union});

  int k = req.topn; /* vector_manager.cc:955: depth = topn when merging */
  if (std::max(k, recall_num) > 1024)
    return err_status(1, "topN (or recall_num) = " +
                             std::to_string(std::max(k, recall_num)) +
                             " exceeds this engine's supported maximum "
                             "of 1024");
  std::vector<double> scores((size_t)nq * k);
  std::vector<int64_t> ids((size_t)nq * k);
  std::string err;
  auto rlock = e->read_lock();
  int rc = e->search_multi(nq, queries, k, nprobe, recall_num, metric,
                           req.brute == 1, req.request_id,
                           req.partition_id, weights,
                           req.multi_vector_rank != 0, scores.data(),
                           ids.data(), &err, &terms, &ranges, req.op,
                           /*prelocked=*/true);
  if (rc == -2) return err_status(-2, "request killed");
  if (rc != 0)
    return err_status(1, err.empty() ? "search failed" : err);

  int nres = req.req_num > 0 ? std::min(req.req_num, nq) : nq;
  std::vector<gpb::SearchResult> results(nres);
  int64_t total = e->num_docs() - e->bitmap().popcount();
  for (int i = 0; i < nres; i++) {
    gpb::SearchResult &res = results[i];
    res.total = (int)total;
    for (int j = req.offset; j < k; j++) {
      int64_t id = ids[(size_t)i * k + j];
      if (id < 0) continue;
      gpb::ResultItem item;
      item.score = scores[(size_t)i * k + j];
      item.fields.push_back({"_id", e->pkey_of(id)});
      for (auto &fname : req.fields) {
        if (fname == "_id") continue;
        if (e->vec_dim_of(fname) > 0) continue;
        const std::string *v = e->field_value(id, fname);
        if (v) item.fields.push_back({fname, *v});
      }
      res.max_score = std::max(res.max_score, item.score);
      res.items.push_back(std::move(item));
    }
  }
  std::string out = gpb::encode_search_response(results);
  *response_str = (char *)malloc(out.size());
  memcpy(*response_str, out.data(), out.size());
  *res_len = (int)out.size();
  return ok_status();
}

static struct CStatus Search_unguarded(void *engine, const char *request_str, int req_len,
                      char **response_str, int *res_len) {
  if (!engine) return err_status(1, "null engine");
  if (memory_exceeded())
    return err_status(-2, "memory limit exceeded"); /* reader.go:170 */
  auto *e = static_cast<Engine *>(engine);
  gpb::SearchRequest req;
  if (!req.parse(request_str, req_len))
    return err_status(1, "parse search request failed");
  if (req.vec_fields.empty())
    return err_status(1, "no vector query (scalar-only search via Query)");
  if (req.topn <= 0) return err_status(1, "limit[topN] is zero");
  if (req.vec_fields.size() > 1)
    return SearchMulti_unguarded(e, req, response_str, res_len);

  const gpb::VectorQuery &vq = req.vec_fields[0];
  int d = e->dimension();
  int nq = (int)(vq.value.size() / ((size_t)d * 4));
  if (nq <= 0) return err_status(1, "Search n shouldn't less than 0!");

  /* retrieval params JSON (ivfpq.cc:233-294): nprobe, recall_num,
   * metric_type, parallel_on_queries (N/A on GPU) */
  int nprobe = 0, recall_num = 0, metric = 0;
  if (!req.index_params.empty()) {
    gjson::Value v;
    if (gjson::parse(req.index_params, v)) {
      v.get_int("nprobe", nprobe);
      v.get_int("recall_num", recall_num);
      std::string mt;
      if (v.get_str("metric_type", mt)) {
        /* only the metric names the reference knows (ivfpq.cc:253-257,
         * DistanceMetricType): anything else is a parse error, not a
         * silent InnerProduct */
        if (strcasecmp(mt.c_str(), "L2") == 0) metric = 1;
        else if (strcasecmp(mt.c_str(), "InnerProduct") == 0 ||
                 strcasecmp(mt.c_str(), "IP") == 0)
          metric = 2;
        else
          return err_status(1, ("unknown metric_type: " + mt).c_str());
      }
    }
  }
  /* scalar filters -> exclusion bitmap consumed by the scan kernels
   * (linear predicate pass, scalar indexes are a later
   * row). filter_operator OR is not supported yet. */
  std::vector<vgamma::TermFilterSpec> terms;
  for (auto &t : req.term_filters)
    terms.push_back({t.field, t.value, t.is_union});
  std::vector<vgamma::RangeFilterSpec> ranges;
  for (auto &t : req.range_filters)
    ranges.push_back({t.field, t.lower, t.upper, t.include_lower,
                      t.include_upper, t.is_union});

  int k = req.topn + req.offset;
  std::vector<float> dists((size_t)nq * k);
  std::vector<int64_t> ids((size_t)nq * k);
  std::string ferr;
  /* hold the shared lock across search AND response assembly so the
   * doc/table state read below (pkey_of, field_value, raw().host_row,
   * bitmap) cannot be reallocated by a concurrent AddOrUpdateDoc /
   * BuildIndex (gamma_api.h threading contract) */
  auto rlock = e->read_lock();
  int rc = e->search(nq, (const float *)vq.value.data(), k, nprobe,
                     recall_num, metric, req.brute == 1, req.request_id,
                     req.partition_id, dists.data(), ids.data(),
                     req.l2_sqrt, &terms, &ranges, &ferr, req.op,
                     /*prelocked=*/true);
  if (rc == -2) return err_status(-2, "request killed");
  if (rc == -3) return err_status(1, ferr);
  if (rc != 0) return err_status(1, "search failed");

  /* score-range filter (SearchCondition::IsSimilarScoreValid,
   * gamma_common_data.h:94) applied on the final top-k */
  bool has_range = vq.has_min || vq.has_max;

  /* the response carries req_num results (response.cc:257 loop); the
   * engine computed nq = bytes/(4d) of them */
  int nres = req.req_num > 0 ? std::min(req.req_num, nq) : nq;
  std::vector<gpb::SearchResult> results(nres);
  int64_t total = e->num_docs() - e->bitmap().popcount();
  bool want_vec = req.is_vector_value;
  for (int i = 0; i < nres; i++) {
    gpb::SearchResult &res = results[i];
    res.total = (int)total;
    for (int j = req.offset; j < k; j++) {
      int64_t id = ids[(size_t)i * k + j];
      if (id < 0) continue;
      double score = dists[(size_t)i * k + j];
      if (has_range && (score < vq.min_score || score > vq.max_score))
        continue;
      gpb::ResultItem item;
      item.score = score;
      item.fields.push_back({"_id", e->pkey_of(id)});
      for (auto &fname : req.fields) {
        if (fname == "_id") continue;
        if (fname == e->vec_field_name()) continue;
        const std::string *v = e->field_value(id, fname);
        if (v) item.fields.push_back({fname, *v});
      }
      if (want_vec) {
        const float *vec = e->raw().host_row(id);
        item.fields.push_back(
            {e->vec_field_name(),
             std::string((const char *)vec, (size_t)d * 4)});
        for (auto &ev : e->extra_vec_fields()) {
          const float *v2 = ev->raw.host_row(id);
          item.fields.push_back(
              {ev->name,
               std::string((const char *)v2, (size_t)ev->dim * 4)});
        }
      }
      res.max_score = std::max(res.max_score, item.score);
      res.items.push_back(std::move(item));
    }
  }
  std::string out = gpb::encode_search_response(results);
  *response_str = (char *)malloc(out.size());
  memcpy(*response_str, out.data(), out.size());
  *res_len = (int)out.size();
  return ok_status();
}

struct CStatus Search(void *engine, const char *request_str, int req_len,
                      char **response_str, int *res_len) {
  try {
    return Search_unguarded(engine, request_str, req_len, response_str, res_len);
  } catch (const std::exception &ex) {
    return err_status(-1,
                      std::string("internal error: ") + ex.what());
  } catch (...) {
    return err_status(-1, "internal error");
  }
}


static struct CStatus Query_unguarded(void *engine, const char *request_str, int req_len,
                     char **response_str, int *res_len) {
  if (!engine) return err_status(1, "null engine");
  auto *e = static_cast<Engine *>(engine);
  gpb::QueryRequest req;
  if (!req.parse(request_str, req_len))
    return err_status(1, "parse query request failed");
  /* stable doc/table state across filtering AND serialization */
  auto rlock = e->read_lock();
  std::vector<int64_t> docids;
  if (!req.document_ids.empty()) {
    for (auto &pk : req.document_ids) {
      int64_t id = e->docid_of(pk);
      if (id >= 0 && !e->bitmap().test(id)) docids.push_back(id);
    }
  } else if (req.n_filters > 0) {
    /* browse by scalar predicate (Engine::Query filter path) */
    std::vector<vgamma::TermFilterSpec> terms;
    for (auto &t : req.term_filters)
      terms.push_back({t.field, t.value, t.is_union});
    std::vector<vgamma::RangeFilterSpec> ranges;
    for (auto &t : req.range_filters)
      ranges.push_back({t.field, t.lower, t.upper, t.include_lower,
                        t.include_upper, t.is_union});
    std::string ferr;
    if (e->filter_docids(terms, ranges, 0,
                         req.limit > 0 ? req.limit : 50, &docids, &ferr,
                         req.op, /*prelocked=*/true))
      return err_status(1, ferr);
  }
  std::vector<gpb::SearchResult> results(1);
  gpb::SearchResult &res = results[0];
  res.total = (int)docids.size();
  for (int64_t id : docids) {
    gpb::ResultItem item;
    item.score = 0;
    item.fields.push_back({"_id", e->pkey_of(id)});
    for (auto &fname : req.fields) {
      if (fname == "_id" || fname == e->vec_field_name()) continue;
      const std::string *v = e->field_value(id, fname);
      if (v) item.fields.push_back({fname, *v});
    }
    if (req.is_vector_value) {
      const float *vec = e->raw().host_row(id);
      item.fields.push_back(
          {e->vec_field_name(),
           std::string((const char *)vec, (size_t)e->dimension() * 4)});
      for (auto &ev : e->extra_vec_fields()) {
        const float *v2 = ev->raw.host_row(id);
        item.fields.push_back(
            {ev->name,
             std::string((const char *)v2, (size_t)ev->dim * 4)});
      }
    }
    res.items.push_back(std::move(item));
  }
  std::string out = gpb::encode_search_response(results);
  *response_str = (char *)malloc(out.size());
  memcpy(*response_str, out.data(), out.size());
  *res_len = (int)out.size();
  return ok_status();
}

struct CStatus Query(void *engine, const char *request_str, int req_len,
                     char **response_str, int *res_len) {
  try {
    return Query_unguarded(engine, request_str, req_len, response_str, res_len);
  } catch (const std::exception &ex) {
    return err_status(-1,
                      std::string("internal error: ") + ex.what());
  } catch (...) {
    return err_status(-1, "internal error");
  }
}


static int SetConfig_unguarded(void *engine, const char *config_str, int len) {
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->set_config(
      std::string(config_str ? config_str : "", len > 0 ? len : 0));
}

int SetConfig(void *engine, const char *config_str, int len) {
  try {
    return SetConfig_unguarded(engine, config_str, len);
  } catch (...) {
    return -1;
  }
}


static int GetConfig_unguarded(void *engine, char **config_str, int *len) {
  std::string s = engine ? static_cast<Engine *>(engine)->get_config()
                         : std::string("{}");
  *config_str = dup_malloc(s);
  *len = (int)s.size();
  return 0;
}

int GetConfig(void *engine, char **config_str, int *len) {
  try {
    return GetConfig_unguarded(engine, config_str, len);
  } catch (...) {
    return -1;
  }
}


static struct CStatus Backup_unguarded(void *engine, int command) {
  if (!engine) return err_status(1, "null engine");
  auto *e = static_cast<Engine *>(engine);
  std::string err;
  if (e->backup(command, &err)) return err_status(1, err.c_str());
  return ok_status();
}

struct CStatus Backup(void *engine, int command) {
  try {
    return Backup_unguarded(engine, command);
  } catch (const std::exception &ex) {
    return err_status(-1,
                      std::string("internal error: ") + ex.what());
  } catch (...) {
    return err_status(-1, "internal error");
  }
}


static struct CStatus AddFieldIndexWithParams_unguarded(
    void *engine, const char *index_name, int index_name_len,
    const char *const *field_names, const int *field_name_lens,
    int field_name_count, const char *index_type, int index_type_len,
    const char *index_params, int index_params_len) {
  if (!engine) return err_status(1, "null engine");
  auto *e = static_cast<Engine *>(engine);
  std::vector<std::string> fields;
  for (int i = 0; i < field_name_count; i++)
    fields.emplace_back(field_names[i], field_name_lens[i]);
  std::string err;
  if (e->add_field_index(std::string(index_name, index_name_len), fields,
                         std::string(index_type, index_type_len),
                         std::string(index_params, index_params_len),
                         &err))
    return err_status(1, err.c_str());
  return ok_status();
}

struct CStatus AddFieldIndexWithParams(
    void *engine, const char *index_name, int index_name_len,
    const char *const *field_names, const int *field_name_lens,
    int field_name_count, const char *index_type, int index_type_len,
    const char *index_params, int index_params_len) {
  try {
    return AddFieldIndexWithParams_unguarded(engine, index_name, index_name_len, field_names, field_name_lens, field_name_count, index_type, index_type_len, index_params, index_params_len);
  } catch (const std::exception &ex) {
    return err_status(-1,
                      std::string("internal error: ") + ex.what());
  } catch (...) {
    return err_status(-1, "internal error");
  }
}


static struct CStatus RemoveFieldIndex_unguarded(void *engine, const char *index_name,
                                int index_name_len) {
  if (!engine) return err_status(1, "null engine");
  auto *e = static_cast<Engine *>(engine);
  std::string err;
  if (e->remove_field_index(std::string(index_name, index_name_len), &err))
    return err_status(1, err.c_str());
  return ok_status();
}

struct CStatus RemoveFieldIndex(void *engine, const char *index_name,
                                int index_name_len) {
  try {
    return RemoveFieldIndex_unguarded(engine, index_name, index_name_len);
  } catch (const std::exception &ex) {
    return err_status(-1,
                      std::string("internal error: ") + ex.what());
  } catch (...) {
    return err_status(-1, "internal error");
  }
}


void SetMemoryLimitConfig(int memory_limit) {
  g_memory_limit_mb = memory_limit;
}

void SetKillStatus(const char *request_id, int partition_id, int reason) {
  (void)reason;
  KillRegistry::inst().set(request_id ? request_id : "", partition_id);
}

void DeleteKillStatus(const char *request_id, int partition_id) {
  KillRegistry::inst().del(request_id ? request_id : "", partition_id);
}

/* ----------------------------------------------------- bench extensions */

static int GammaBulkAdd_unguarded(void *engine, const char *field, int field_len, int n,
                 const float *vecs) {
  (void)field;
  (void)field_len;
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->bulk_add(n, vecs);
}

int GammaBulkAdd(void *engine, const char *field, int field_len, int n,
                 const float *vecs) {
  try {
    return GammaBulkAdd_unguarded(engine, field, field_len, n, vecs);
  } catch (...) {
    return -1;
  }
}


static int GammaRawSearch_unguarded(void *engine, int nq, const float *xq, int k, int nprobe,
                   int rerank, int metric, float *out_dists,
                   int64_t *out_ids) {
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->search(
      nq, xq, k, nprobe, rerank, metric, false, "", 0, out_dists, out_ids);
}

int GammaRawSearch(void *engine, int nq, const float *xq, int k, int nprobe,
                   int rerank, int metric, float *out_dists,
                   int64_t *out_ids) {
  try {
    return GammaRawSearch_unguarded(engine, nq, xq, k, nprobe, rerank, metric, out_dists, out_ids);
  } catch (...) {
    return -1;
  }
}


static int GammaCacheQueries_unguarded(void *engine, int nq, const float *xq) {
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->cache_queries(nq, xq);
}

int GammaCacheQueries(void *engine, int nq, const float *xq) {
  try {
    return GammaCacheQueries_unguarded(engine, nq, xq);
  } catch (...) {
    return -1;
  }
}


static int GammaRawSearchCached_unguarded(void *engine, int nq, int k, int nprobe,
                         int rerank, int metric, float *out_dists,
                         int64_t *out_ids) {
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->search(
      nq, nullptr, k, nprobe, rerank, metric, false, "", 0, out_dists,
      out_ids);
}

int GammaRawSearchCached(void *engine, int nq, int k, int nprobe,
                         int rerank, int metric, float *out_dists,
                         int64_t *out_ids) {
  try {
    return GammaRawSearchCached_unguarded(engine, nq, k, nprobe, rerank, metric, out_dists, out_ids);
  } catch (...) {
    return -1;
  }
}


int GammaDebugCoarseAssign(void *engine, int nq, const float *xq,
                           int nprobe, int64_t *out_lists,
                           float *out_dists) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  auto *ix = e->index();
  if (!ix || !ix->trained()) return -1;
  hipStream_t s = e->stream();
  vgamma::DeviceBuf qd, qn, probes, pdists;
  int d = e->dimension();
  if (qd.reserve((size_t)nq * d * 4)) return -1;
  (void)hipMemcpy(qd.get(), xq, (size_t)nq * d * 4, hipMemcpyHostToDevice);
  if (qn.reserve((size_t)nq * 4)) return -1;
  (void)gk::row_norms(s, qd.as<float>(), nq, d, qn.as<float>());
  if (probes.reserve((size_t)nq * nprobe * 8)) return -1;
  if (pdists.reserve((size_t)nq * nprobe * 4)) return -1;
  vgamma::SearchScratch tmp_sc; /* debug path: throwaway scratch */
  tmp_sc.stream = s;
  if (ix->coarse_assign(qd.as<float>(), nq, nprobe,
                        e->metric_ip_default(), qn.as<float>(), s,
                        probes.as<int64_t>(), pdists.as<float>(),
                        tmp_sc)) {
    tmp_sc.stream = nullptr; /* not ours to destroy */
    return -1;
  }
  tmp_sc.stream = nullptr; /* not ours to destroy */
  (void)hipStreamSynchronize(s);
  (void)hipMemcpy(out_lists, probes.get(), (size_t)nq * nprobe * 8,
            hipMemcpyDeviceToHost);
  (void)hipMemcpy(out_dists, pdists.get(), (size_t)nq * nprobe * 4,
            hipMemcpyDeviceToHost);
  return 0;
}

int GammaDebugGetOPQ(void *engine, float *R) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  auto *ix = e->index();
  if (!ix || !ix->has_opq() || ix->opq_R_host().empty()) return -1;
  memcpy(R, ix->opq_R_host().data(), ix->opq_R_host().size() * 4);
  return 0;
}

int GammaDebugApplyOPQ(void *engine, const float *xq, int nq, float *out) {
  if (!engine || nq <= 0) return -1;
  auto *e = static_cast<Engine *>(engine);
  auto *ix = e->index();
  if (!ix || !ix->has_opq()) return -1;
  hipStream_t s = e->stream();
  int d = e->dimension();
  vgamma::DeviceBuf qd, qr;
  if (qd.reserve((size_t)nq * d * 4) || qr.reserve((size_t)nq * d * 4))
    return -1;
  (void)hipMemcpy(qd.get(), xq, (size_t)nq * d * 4, hipMemcpyHostToDevice);
  if (ix->rotate_dev(qd.as<float>(), nq, qr.as<float>(), s)) return -1;
  (void)hipStreamSynchronize(s);
  (void)hipMemcpy(out, qr.get(), (size_t)nq * d * 4, hipMemcpyDeviceToHost);
  return 0;
}

int GammaDebugGetModel(void *engine, float *centroids, float *codebooks) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  if (!e->index()) return -1;
  return e->index()->copy_model_to_host(centroids, codebooks, e->stream());
}

int64_t GammaDebugGetList(void *engine, int64_t list_no, int64_t *ids,
                          uint8_t *codes) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  if (!e->index()) return -1;
  int64_t sz = e->index()->list_size(list_no);
  if (sz < 0) return -1;
  if (ids || codes)
    if (e->index()->copy_list_to_host(list_no, ids, codes, e->stream()))
      return -1;
  return sz;
}

int64_t GammaDebugNumDocs(void *engine) {
  if (!engine) return -1;
  return static_cast<Engine *>(engine)->num_docs();
}

int GammaLastSearchTiming(void *engine, double *us6) {
  if (!engine) return -1;
  auto *e = static_cast<Engine *>(engine);
  for (int i = 0; i < 6; i++) us6[i] = e->last_timing[i];
  return 0;
}

/* ------------------------------------------- codec self-tests (CPU-only)
 * Used by tests/test_capi_cpu.py to exercise the hand-written protobuf /
 * FlatBuffers codecs without a GPU. */

static int GammaTestParseSearchRequest_unguarded(const char *buf, int len, char **json_out,
                                int *json_len) {
  gpb::SearchRequest req;
  if (!req.parse(buf, len)) return -1;
  std::string s = "{";
  s += "\"request_id\": \"" + gjson::escape(req.request_id) + "\",";
  s += "\"partition_id\": " + std::to_string(req.partition_id) + ",";
  s += "\"req_num\": " + std::to_string(req.req_num) + ",";
  s += "\"topn\": " + std::to_string(req.topn) + ",";
  s += "\"brute\": " + std::to_string(req.brute) + ",";
  s += "\"n_vec\": " + std::to_string(req.vec_fields.size()) + ",";
  if (!req.vec_fields.empty()) {
    s += "\"vec_name\": \"" + gjson::escape(req.vec_fields[0].name) + "\",";
    s += "\"vec_bytes\": " +
         std::to_string(req.vec_fields[0].value.size()) + ",";
    s += "\"min_score\": " + std::to_string(req.vec_fields[0].min_score) +
         ",";
    s += "\"max_score\": " + std::to_string(req.vec_fields[0].max_score) +
         ",";
  }
  s += "\"index_params\": \"" + gjson::escape(req.index_params) + "\",";
  s += "\"l2_sqrt\": " + std::to_string(req.l2_sqrt ? 1 : 0) + ",";
  s += "\"n_fields\": " + std::to_string(req.fields.size()) + ",";
  s += "\"op\": " + std::to_string(req.op) + ",";
  s += "\"term_unions\": [";
  for (size_t i = 0; i < req.term_filters.size(); i++)
    s += (i ? "," : "") + std::to_string(req.term_filters[i].is_union);
  s += "],";
  s += "\"range_unions\": [";
  for (size_t i = 0; i < req.range_filters.size(); i++)
    s += (i ? "," : "") + std::to_string(req.range_filters[i].is_union);
  s += "],";
  s += "\"n_filters\": " + std::to_string(req.n_filters) + "}";
  *json_out = dup_malloc(s);
  *json_len = (int)s.size();
  return 0;
}

int GammaTestParseSearchRequest(const char *buf, int len, char **json_out,
                                int *json_len) {
  try {
    return GammaTestParseSearchRequest_unguarded(buf, len, json_out, json_len);
  } catch (...) {
    return -1;
  }
}


static int GammaTestParseTable_unguarded(const char *buf, int len, char **json_out,
                        int *json_len) {
  gfb::TableSchema ts;
  if (!ts.parse(buf, (size_t)len)) return -1;
  std::string s = "{";
  s += "\"name\": \"" + gjson::escape(ts.name) + "\",";
  s += "\"index_type\": \"" + gjson::escape(ts.index_type) + "\",";
  s += "\"index_params\": \"" + gjson::escape(ts.index_params) + "\",";
  s += "\"n_fields\": " + std::to_string(ts.fields.size()) + ",";
  s += "\"n_vectors\": " + std::to_string(ts.vectors.size()) + ",";
  if (!ts.vectors.empty()) {
    s += "\"vec_name\": \"" + gjson::escape(ts.vectors[0].name) + "\",";
    s += "\"dimension\": " + std::to_string(ts.vectors[0].dimension) + ",";
  }
  std::string fn;
  for (auto &f : ts.fields) fn += f.name + ",";
  s += "\"field_names\": \"" + gjson::escape(fn) + "\"}";
  *json_out = dup_malloc(s);
  *json_len = (int)s.size();
  return 0;
}

int GammaTestParseTable(const char *buf, int len, char **json_out,
                        int *json_len) {
  try {
    return GammaTestParseTable_unguarded(buf, len, json_out, json_len);
  } catch (...) {
    return -1;
  }
}


static int GammaTestDocRoundtrip_unguarded(const char *buf, int len, char **out,
                          int *out_len) {
  gfb::Doc doc;
  if (!doc.parse(buf, (size_t)len)) return -1;
  std::string s = doc.serialize();
  gfb::Doc doc2;
  if (!doc2.parse(s.data(), s.size())) return -2;
  if (doc2.fields.size() != doc.fields.size()) return -3;
  for (size_t i = 0; i < doc.fields.size(); i++) {
    if (doc.fields[i].name != doc2.fields[i].name) return -4;
    if (doc.fields[i].value != doc2.fields[i].value) return -5;
    if (doc.fields[i].data_type != doc2.fields[i].data_type) return -6;
  }
  *out = (char *)malloc(s.size());
  memcpy(*out, s.data(), s.size());
  *out_len = (int)s.size();
  return 0;
}

int GammaTestDocRoundtrip(const char *buf, int len, char **out,
                          int *out_len) {
  try {
    return GammaTestDocRoundtrip_unguarded(buf, len, out, out_len);
  } catch (...) {
    return -1;
  }
}


} /* extern "C" */
