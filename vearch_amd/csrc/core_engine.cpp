/*
 * core_engine.cpp — Engine: the thin reimplementation of
 * search/engine.cc + vector/vector_manager.cc above the index models.
 */
#include <math.h>
#include <sys/stat.h>

#include <algorithm>

#include "core.hpp"
#include "json.hpp"

namespace vgamma {

static const std::string kEmpty;

int Engine::init(const std::string &config_json, std::string *err) {
  /* engine config JSON: path, log_dir, space_name (gamma_api.cc:36-70) */
  gjson::Value v;
  if (!config_json.empty() && !gjson::parse(config_json, v)) {
    if (err) *err = "bad engine config json";
    return -1;
  }
  v.get_str("path", path_);
  v.get_str("log_dir", log_dir_);
  v.get_str("space_name", space_name_);
  if (path_.empty()) path_ = ".";
  int dev_count = 0;
  if (hipGetDeviceCount(&dev_count) != hipSuccess || dev_count == 0) {
    if (err) *err = "no HIP device: the MI355X engine requires a GPU";
    return -1;
  }
  GAMMA_CHECK(hipStreamCreate(&stream_));
  return 0;
}

bool Engine::metric_ip_default() const { return params_.metric_ip; }

int Engine::create_table(const std::string &name,
                         const std::vector<FieldMeta> &scalar_fields,
                         const std::string &vec_name, int dimension,
                         const std::string &index_type,
                         const std::string &index_params_json,
                         int training_threshold, std::string *err,
                         const std::vector<std::pair<std::string, int>>
                             &extra_vec_fields) {
  if (table_created_) {
    if (err) *err = "table already created";
    return -1;
  }
  if (dimension <= 0 || dimension % 4 != 0) {
    if (err) *err = "dimension must be a positive multiple of 4";
    return -1;
  }
  space_name_ = space_name_.empty() ? name : space_name_;
  fields_ = scalar_fields;
  for (auto &f : fields_) field_vals_[f.name]; /* default-construct */
  vec_name_ = vec_name;
  dim_ = dimension;
  index_type_ = index_type.empty() ? "IVFPQ" : index_type;

  if (params_.parse(index_params_json, err)) return -1;
  if (index_type_ == "IVFPQ") params_.kind = IndexKind::IVFPQ;
  else if (index_type_ == "IVFFLAT") params_.kind = IndexKind::IVFFLAT;
  else if (index_type_ == "FLAT") params_.kind = IndexKind::FLAT;
  else {
    if (err)
      *err = "index type " + index_type_ +
             " not in this build's scope (FLAT/IVFFLAT/IVFPQ)";
    return -1;
  }
  if (raw_.init(dim_)) return -1;

  if (params_.kind != IndexKind::FLAT) {
    index_ = std::make_unique<IVFIndex>();
    if (index_->init(dim_, params_)) {
      if (err) *err = "index init failed (check nsubvector divides d, %4)";
      return -1;
    }
  }
  /* extra vector fields (multi-vector table, vector_manager.cc:898):
   * each gets its own raw store + index of the table's index type;
   * nsubvector falls back to the per-field default when the table's
   * value does not divide the field's dimension */
  for (auto &ev : extra_vec_fields) {
    if (ev.second <= 0 || ev.second % 4 != 0) {
      if (err) *err = "extra vector field " + ev.first + ": bad dim";
      return -1;
    }
    auto e = std::make_unique<ExtraVecField>();
    e->name = ev.first;
    e->dim = ev.second;
    if (e->raw.init(e->dim)) return -1;
    if (params_.kind != IndexKind::FLAT) {
      IndexParams pe = params_;
      if (pe.nsubvector && e->dim % pe.nsubvector) pe.nsubvector = 0;
      e->index = std::make_unique<IVFIndex>();
      if (e->index->init(e->dim, pe)) {
        if (err) *err = "extra vector index init failed for " + ev.first;
        return -1;
      }
    }
    extra_vecs_.push_back(std::move(e));
  }
  /* training_threshold default: max(nlist*39, 256*39) vectors
   * (ivfpq.cc:137-144 with default_points_per_centroid=39) */
  if (training_threshold > 0) training_threshold_ = training_threshold;
  else if (params_.training_threshold > 0)
    training_threshold_ = params_.training_threshold;
  else
    training_threshold_ = std::max(params_.ncentroids * 39, 256 * 39);
  table_created_ = true;
  return 0;
}

int Engine::add_doc(
    const std::string &p_key,
    const std::vector<std::pair<std::string, std::string>> &fields,
    const float *vec, int vec_len,
    const std::vector<MultiVecQuery> *extra_vecs) {
  if (!table_created_ || vec_len != dim_) return -1;
  /* every extra vector field of the table must be supplied */
  if (!extra_vecs_.empty()) {
    size_t have = extra_vecs ? extra_vecs->size() : 0;
    if (have != extra_vecs_.size()) return -1;
    for (auto &q : *extra_vecs)
      if (!extra_vec_(q.name) || !q.vecs) return -1;
  }
  /* Lock-free common append (realtime_mem_data.cc:57-68
   * retrieve_idx_pos_ pattern): a pure append that fits every existing
   * capacity runs under the SHARED lock — concurrent searches are
   * never blocked — with all row state written before the new docid /
   * bucket size is published. Anything structural (update of an
   * existing pkey, raw-segment or bucket growth, bitmap growth)
   * retries under the write lock. */
  if (extra_vecs_.empty()) { /* multi-vector rows take the slow path */
    std::shared_lock<std::shared_mutex> g(rw_);
    std::lock_guard<std::mutex> ap(append_mu_);
    int rc = add_doc_fast_(p_key, fields, vec);
    if (rc != 1) return rc;
  }
  std::unique_lock<std::shared_mutex> g(rw_);
  std::lock_guard<std::mutex> ap(append_mu_);
  {
    std::lock_guard<std::mutex> pk(pkey_mu_);
    auto it = pkey2docid_.find(p_key);
    if (it != pkey2docid_.end()) {
      /* update = delete old + add new (engine.cc AddOrUpdate
       * semantics) */
      int64_t old = it->second;
      bitmap_.set(old, stream_);
      if (index_) index_->del(old, stream_);
      for (auto &e : extra_vecs_)
        if (e->index) e->index->del(old, stream_);
    }
  }
  int64_t docid = max_docid_.load(std::memory_order_relaxed);
  if (raw_.size() != docid) return -1; /* see add_doc_fast_ */
  docid2pkey_.set(docid, p_key);
  bitmap_.ensure(docid + 1, stream_);
  for (auto &f : fields_) {
    auto &col = field_vals_[f.name];
    col.resize(docid + 1);
    for (auto &kv : fields)
      if (kv.first == f.name) col[docid] = kv.second;
  }
  if (raw_.add(vec, 1, stream_)) return -1;
  if (index_ && index_->trained()) {
    if (index_->add(vec, &docid, 1, stream_)) return -1;
    indexed_count_++;
  }
  for (auto &e : extra_vecs_) { /* parallel rows, same docid */
    const float *v = nullptr;
    for (auto &q : *extra_vecs)
      if (q.name == e->name) v = q.vecs;
    if (e->raw.add(v, 1, stream_)) return -1;
    if (e->index && e->index->trained() &&
        e->index->add(v, &docid, 1, stream_))
      return -1;
  }
  {
    std::lock_guard<std::mutex> pk(pkey_mu_);
    pkey2docid_[p_key] = docid;
  }
  max_docid_.store(docid + 1, std::memory_order_release);
  return 0;
}

int Engine::add_doc_fast_(
    const std::string &p_key,
    const std::vector<std::pair<std::string, std::string>> &fields,
    const float *vec) {
  {
    std::lock_guard<std::mutex> pk(pkey_mu_);
    if (pkey2docid_.count(p_key)) return 1; /* update -> slow path */
  }
  int64_t docid = max_docid_.load(std::memory_order_relaxed);
  if (docid + 1 > 0x7fffffff) return -1; /* 31-bit device vid */
  if (raw_.size() != docid) return -1; /* row/docid drift (a prior
                                          failed append): fail loudly
                                          rather than mis-map rows */
  if (raw_.would_grow(1)) return 1;
  if (!bitmap_.has_capacity(docid + 1)) return 1;
  int32_t bucket = -1;
  uint8_t code[256];
  float sval = 0.0f;
  bool commit_index = false;
  if (index_ && index_->trained()) {
    if (index_->code_size() > (int)sizeof(code)) return 1;
    int rc = index_->prepare_fast_one(vec, stream_, &bucket, code, &sval);
    if (rc != 0) return rc; /* 1 = bucket needs extension -> slow */
    commit_index = true;
  }
  /* all row state BEFORE publication */
  docid2pkey_.set(docid, p_key);
  for (auto &f : fields_) {
    auto &col = field_vals_[f.name];
    col.resize(docid + 1);
    for (auto &kv : fields)
      if (kv.first == f.name) col[docid] = kv.second;
  }
  if (raw_.add(vec, 1, stream_)) return -1; /* publishes raw n_ last */
  if (commit_index) {
    if (index_->commit_fast_one(bucket, docid, vec, code, sval, stream_))
      return -1;
    indexed_count_++;
  }
  {
    std::lock_guard<std::mutex> pk(pkey_mu_);
    pkey2docid_[p_key] = docid;
  }
  max_docid_.store(docid + 1, std::memory_order_release);
  return 0;
}

int Engine::bulk_add(int64_t n, const float *vecs) {
  if (!table_created_ || n <= 0) return -1;
  std::unique_lock<std::shared_mutex> g(rw_);
  int64_t base = max_docid_.load(std::memory_order_relaxed);
  if ((uint64_t)(base + n) > 0x7fffffffull) return -1; /* 31-bit vids */
  char buf[24];
  {
    std::lock_guard<std::mutex> pk(pkey_mu_);
    for (int64_t i = 0; i < n; i++) {
      int64_t docid = base + i;
      snprintf(buf, sizeof buf, "%lld", (long long)docid);
      pkey2docid_.emplace(buf, docid);
      docid2pkey_.set(docid, buf);
    }
  }
  max_docid_.store(base + n, std::memory_order_release);
  bitmap_.ensure(base + n, stream_);
  if (raw_.add(vecs, n, stream_)) return -1;
  if (index_ && index_->trained()) {
    std::vector<int64_t> vids(n);
    for (int64_t i = 0; i < n; i++) vids[i] = base + i;
    if (index_->add(vecs, vids.data(), n, stream_)) return -1;
    indexed_count_ += n;
  }
  return 0;
}

int Engine::delete_doc(const std::string &p_key) {
  std::unique_lock<std::shared_mutex> g(rw_);
  std::lock_guard<std::mutex> pk(pkey_mu_);
  auto it = pkey2docid_.find(p_key);
  if (it == pkey2docid_.end()) return -1;
  bitmap_.set(it->second, stream_);
  if (index_) index_->del(it->second, stream_);
  for (auto &e : extra_vecs_)
    if (e->index) e->index->del(it->second, stream_);
  pkey2docid_.erase(it);
  return 0;
}

int Engine::build_index(std::string *err) {
  if (!table_created_) {
    if (err) *err = "no table";
    return -1;
  }
  if (params_.kind == IndexKind::FLAT) return 0;
  std::unique_lock<std::shared_mutex> g(rw_);
  if (index_->trained()) return 0;
  /* training size clamp (ivfpq.cc:296-329): [nlist*39, nlist*256] */
  int64_t nlist = params_.ncentroids;
  int64_t num = training_threshold_;
  if (num < nlist) num = nlist * 39;
  else if (num > nlist * 256) num = nlist * 256;
  else if (num < nlist * 39) { /* warning case, keep as is */
  }
  if (num > raw_.size()) {
    if (err)
      *err = "vector count " + std::to_string(raw_.size()) +
             " less than training threshold " + std::to_string(num);
    return -1;
  }
  std::vector<float> xt((size_t)num * dim_);
  raw_.host_copy(0, num, xt.data());
  if (index_->train(xt.data(), num, stream_, err)) return -1;
  /* index every existing vector (Indexing/AddRTVecsToIndex loop,
   * engine.cc:1084-1127) */
  const int64_t chunk = 262144;
  std::vector<float> buf;
  std::vector<int64_t> vids;
  for (int64_t c0 = 0; c0 < raw_.size(); c0 += chunk) {
    int64_t cn = std::min(chunk, raw_.size() - c0);
    buf.resize((size_t)cn * dim_);
    raw_.host_copy(c0, cn, buf.data());
    vids.resize(cn);
    for (int64_t i = 0; i < cn; i++) vids[i] = c0 + i;
    if (index_->add(buf.data(), vids.data(), cn, stream_)) return -1;
  }
  indexed_count_ = raw_.size();
  /* extra vector fields: train + index each on its own raw store */
  for (auto &e : extra_vecs_) {
    if (!e->index || e->index->trained()) continue;
    int64_t enum_ = std::min<int64_t>(num, e->raw.size());
    std::vector<float> ext((size_t)enum_ * e->dim);
    e->raw.host_copy(0, enum_, ext.data());
    if (e->index->train(ext.data(), enum_, stream_, err)) return -1;
    for (int64_t c0 = 0; c0 < e->raw.size(); c0 += chunk) {
      int64_t cn = std::min(chunk, e->raw.size() - c0);
      buf.resize((size_t)cn * e->dim);
      e->raw.host_copy(c0, cn, buf.data());
      vids.resize(cn);
      for (int64_t i = 0; i < cn; i++) vids[i] = c0 + i;
      if (e->index->add(buf.data(), vids.data(), cn, stream_)) return -1;
    }
  }
  return 0;
}

int Engine::rebuild_index(bool drop_before_rebuild, std::string *err) {
  if (!table_created_) {
    if (err) *err = "no table";
    return -1;
  }
  if (params_.kind == IndexKind::FLAT) return 0;
  {
    std::unique_lock<std::shared_mutex> g(rw_);
    if (drop_before_rebuild || index_->trained()) {
      index_ = std::make_unique<IVFIndex>();
      if (index_->init(dim_, params_)) {
        if (err) *err = "index re-init failed";
        return -1;
      }
      indexed_count_ = 0;
      for (auto &e : extra_vecs_) { /* extras retrain with the table */
        if (!e->index) continue;
        IndexParams pe = params_;
        if (pe.nsubvector && e->dim % pe.nsubvector) pe.nsubvector = 0;
        e->index = std::make_unique<IVFIndex>();
        if (e->index->init(e->dim, pe)) {
          if (err) *err = "extra index re-init failed";
          return -1;
        }
      }
    }
  }
  return build_index(err);
}

int Engine::flat_search_keys(RawStore &raw, int dim, const float *q_dev,
                             int nq, int k2, const float *q_norms_dev,
                             bool ip, hipStream_t s,
                             uint64_t *out_keys_dev, const uint32_t *bm,
                             SearchScratch &sc) {
  const int64_t n = raw.size();
  if (nq < 512 || n < 200000) {
    GAMMA_CHECK(gk::flat_stream_scan(
        s, nq, n, dim, k2, q_dev, raw.dev_seg_table(),
        RawStore::SEG_SHIFT, bm, ip, out_keys_dev));
    return 0;
  }
  /* chunked MFMA GEMM + seeded select (nq large): per segment-run chunks */
  const int64_t chunk = 65536;
  if (sc.flat_dots.reserve((size_t)nq * chunk * 4)) return -1;
  bool seeded = false;
  for (int64_t v0 = 0; v0 < n;) {
    int64_t run = 0;
    const float *seg = raw.dev_run(v0, &run);
    int64_t take = std::min(run, chunk);
    GAMMA_CHECK(gk::dots_mfma(s, q_dev, nq, seg, take, dim,
                              sc.flat_dots.as<float>()));
    GAMMA_CHECK(gk::select_from_dots(
        s, nq, take, v0, take, sc.flat_dots.as<float>(), q_norms_dev,
        raw.dev_norms(), !ip, ip, bm, k2, out_keys_dev, seeded));
    seeded = true;
    v0 += take;
  }
  return 0;
}

namespace {
/* decode a raw-binary scalar by gamma DataType for range compares */
static bool decode_num(int dt, const std::string &b, double *out) {
  if (dt == 0 && b.size() >= 4) { /* INT */
    int32_t v; memcpy(&v, b.data(), 4); *out = v; return true;
  }
  if ((dt == 1 || dt == 7) && b.size() >= 8) { /* LONG / DATE */
    int64_t v; memcpy(&v, b.data(), 8); *out = (double)v; return true;
  }
  if (dt == 2 && b.size() >= 4) { /* FLOAT */
    float v; memcpy(&v, b.data(), 4); *out = v; return true;
  }
  if (dt == 3 && b.size() >= 8) { /* DOUBLE */
    double v; memcpy(&v, b.data(), 8); *out = v; return true;
  }
  if (dt == 6 && b.size() >= 1) { /* BOOL */
    *out = b[0] != 0; return true;
  }
  return false;
}

/* \x01-separated multi-values: filter terms and STRINGARRAY doc
 * values both use this encoding (c_api/api_data/doc.cc:102) */
static std::vector<std::string> split_x01(const std::string &s) {
  std::vector<std::string> out;
  size_t p = 0;
  while (p <= s.size()) {
    size_t q = s.find('\x01', p);
    if (q == std::string::npos) { out.push_back(s.substr(p)); break; }
    out.push_back(s.substr(p, q - p));
    p = q + 1;
  }
  return out;
}
}  // namespace

/* Append rows [ix.*_upto, upto) of field `fname` into its scalar index
 * (term postings and/or value-sorted arrays). Caller holds scalar_mu_.
 * The lazy path (filtered search) and the explicit path
 * (AddFieldIndexWithParams) share this. */
const ScalarFieldIndex *Engine::extend_scalar_index_(
    const std::string &fname, int dt, bool want_terms, bool want_range,
    int64_t upto) {
  auto cit = field_vals_.find(fname);
  if (cit == field_vals_.end()) return nullptr;
  const StableStrCol &col = cit->second;
  ScalarFieldIndex &ix = scalar_idx_[fname];
  static const std::string kNone;
  const int64_t n = upto;
  if (want_terms && ix.terms_upto < n) {
    for (int64_t id = ix.terms_upto; id < n; id++) {
      const std::string &v = id < (int64_t)col.size() ? col[id] : kNone;
      if (dt == 8) {
        for (auto &el : split_x01(v))
          if (!el.empty()) ix.postings[el].push_back(id);
      } else {
        ix.postings[v].push_back(id);
      }
    }
    ix.terms_upto = n;
  }
  if (want_range && ix.range_upto < n) {
    for (int64_t id = ix.range_upto; id < n; id++) {
      const std::string &v = id < (int64_t)col.size() ? col[id] : kNone;
      if (dt == 4) {
        ix.svals.emplace_back(v, id);
      } else {
        double x;
        if (decode_num(dt, v, &x)) ix.nvals.emplace_back(x, id);
        /* decode failure: absent from nvals -> fails the range */
      }
    }
    ix.range_upto = n;
    std::sort(ix.svals.begin(), ix.svals.end());
    std::sort(ix.nvals.begin(), ix.nvals.end());
  }
  return &ix;
}

int Engine::build_filter_bitmap_(const std::vector<TermFilterSpec> &terms,
                                 const std::vector<RangeFilterSpec> &ranges,
                                 SearchScratch &sc,
                                 const uint32_t **dev_out,
                                 std::string *err, int filter_op) {
  const int64_t n = max_docid_;
  const int64_t words = (n + 31) / 32;
  /* resolve fields + lazily extend their scalar indexes. Semantics are
   * identical to a per-doc scan of the raw column bytes: a doc is
   * included iff it is live AND matches every filter; docs whose bytes
   * fail numeric decode (or rows missing a value) fail that filter. */
  struct Q { const ScalarFieldIndex *ix; int dt;
             const TermFilterSpec *tf; const RangeFilterSpec *rf;
             double lo, hi; };
  std::vector<Q> qs;
  {
    std::lock_guard<std::mutex> lk(scalar_mu_);
    auto dtype_of = [&](const std::string &f) {
      for (auto &fm : fields_)
        if (fm.name == f) return fm.data_type;
      return 0;
    };
    auto extend = [&](const std::string &fname, int dt, bool want_terms,
                      bool want_range) -> const ScalarFieldIndex * {
      return extend_scalar_index_(fname, dt, want_terms, want_range, n);
    };
    for (auto &t : terms) {
      int dt = dtype_of(t.field);
      const ScalarFieldIndex *ix = extend(t.field, dt, true, false);
      if (!ix) {
        if (err) *err = "unknown filter field " + t.field;
        return -1;
      }
      qs.push_back({ix, dt, &t, nullptr, 0, 0});
    }
    for (auto &r : ranges) {
      int dt = dtype_of(r.field);
      const ScalarFieldIndex *ix = extend(r.field, dt, false, true);
      if (!ix) {
        if (err) *err = "unknown filter field " + r.field;
        return -1;
      }
      Q q{ix, dt, nullptr, &r, 0, 0};
      if (dt != 4 && dt != 8) {
        if (!decode_num(dt, r.lower, &q.lo) ||
            !decode_num(dt, r.upper, &q.hi)) {
          if (err) *err = "bad range filter value for field " + r.field;
          return -1;
        }
      }
      qs.push_back(q);
    }
  }

  /* per-filter match sets, combined by the request-level FilterOperator
   * (scalar_index_manager.cc:1188-1190: And -> Intersection,
   * Or -> Union). Per-filter is_union follows engine.cc:475 /
   * scalar_index_types.h:44 (And=0, Or=1, Not=2): a term filter with
   * Not takes the complement of the matched set over [0, maxdoc)
   * (BitmapIndex::NotIn, bitmap_index.cc:133 — docs missing the field
   * DO match Not); a numeric range filter with Not and equal inclusive
   * bounds is NotEqual (bitmap_index.cc:196). */
  const bool op_or = filter_op == 1;
  std::vector<uint32_t> inc((size_t)words, op_or ? 0u : 0xffffffffu);
  std::vector<uint32_t> mt;
  for (auto &q : qs) {
    mt.assign((size_t)words, 0);
    bool negate = false;
    if (q.tf) {
      negate = q.tf->is_union == 2;
      for (auto &t : split_x01(q.tf->value)) {
        auto pit = q.ix->postings.find(t);
        if (pit == q.ix->postings.end()) continue;
        for (int64_t id : pit->second)
          if (id < n) mt[id >> 5] |= 1u << (id & 31);
      }
    } else if (q.rf->is_union == 2 && q.dt != 4 && q.dt != 8 &&
               q.rf->lower == q.rf->upper && q.rf->inc_l && q.rf->inc_u) {
      /* numeric NotEqual: complement of Equal(lower) */
      negate = true;
      const auto &nv = q.ix->nvals;
      auto b = std::lower_bound(
          nv.begin(), nv.end(), q.lo,
          [](const std::pair<double, int64_t> &a, double v) {
            return a.first < v;
          });
      for (; b < nv.end() && b->first == q.lo; ++b)
        if (b->second < n) mt[b->second >> 5] |= 1u << (b->second & 31);
    } else if (q.dt == 4) { /* lexicographic string range */
      const auto &sv = q.ix->svals;
      auto cmp = [](const std::pair<std::string, int64_t> &a,
                    const std::string &b) { return a.first < b; };
      auto cmp2 = [](const std::string &a,
                     const std::pair<std::string, int64_t> &b) {
        return a < b.first;
      };
      auto b = q.rf->inc_l
                   ? std::lower_bound(sv.begin(), sv.end(), q.rf->lower, cmp)
                   : std::upper_bound(sv.begin(), sv.end(), q.rf->lower,
                                      cmp2);
      auto e = q.rf->inc_u
                   ? std::upper_bound(sv.begin(), sv.end(), q.rf->upper,
                                      cmp2)
                   : std::lower_bound(sv.begin(), sv.end(), q.rf->upper,
                                      cmp);
      for (; b < e; ++b)
        if (b->second < n) mt[b->second >> 5] |= 1u << (b->second & 31);
    } else if (q.dt != 8) { /* numeric range (range on STRINGARRAY
                               matches nothing, as in the byte-scan) */
      const auto &nv = q.ix->nvals;
      auto cmp = [](const std::pair<double, int64_t> &a, double b) {
        return a.first < b;
      };
      auto cmp2 = [](double a, const std::pair<double, int64_t> &b) {
        return a < b.first;
      };
      auto b = q.rf->inc_l
                   ? std::lower_bound(nv.begin(), nv.end(), q.lo, cmp)
                   : std::upper_bound(nv.begin(), nv.end(), q.lo, cmp2);
      auto e = q.rf->inc_u
                   ? std::upper_bound(nv.begin(), nv.end(), q.hi, cmp2)
                   : std::lower_bound(nv.begin(), nv.end(), q.hi, cmp);
      for (; b < e; ++b)
        if (b->second < n) mt[b->second >> 5] |= 1u << (b->second & 31);
    }
    if (negate) {
      /* complement over [0, n): flip, then clear the tail bits past n */
      for (int64_t w = 0; w < words; w++) mt[w] = ~mt[w];
      if (n & 31) mt[words - 1] &= (1u << (n & 31)) - 1;
    }
    if (op_or)
      for (int64_t w = 0; w < words; w++) inc[w] |= mt[w];
    else
      for (int64_t w = 0; w < words; w++) inc[w] &= mt[w];
  }

  sc.filt_host.assign((size_t)words, 0);
  for (int64_t w = 0; w < words; w++)
    sc.filt_host[w] = ~inc[w] | bitmap_.host_word(w);
  if (!dev_out) return 0; /* browse path: host bitmap only */
  if (sc.filt_dev.reserve((size_t)std::max<int64_t>(words, 1) * 4))
    return -1;
  GAMMA_CHECK(hipMemcpyAsync(sc.filt_dev.get(), sc.filt_host.data(),
                             (size_t)words * 4, hipMemcpyHostToDevice,
                             sc.stream));
  *dev_out = sc.filt_dev.as<uint32_t>();
  return 0;
}

SearchScratch *Engine::acquire_scratch_() {
  std::unique_lock<std::mutex> lk(pool_mu_);
  for (;;) {
    for (auto &p : pool_)
      if (!p->in_use) {
        p->in_use = true;
        return p.get();
      }
    if ((int)pool_.size() < kMaxConcurrentSearches) {
      auto p = std::make_unique<SearchScratch>();
      if (hipStreamCreate(&p->stream) != hipSuccess) return nullptr;
      p->in_use = true;
      pool_.push_back(std::move(p));
      return pool_.back().get();
    }
    pool_cv_.wait(lk);
  }
}

void Engine::release_scratch_(SearchScratch *sc) {
  /* all async work on this context must be done before another caller
   * (or a write-locked mutator) can touch shared device state */
  (void)hipStreamSynchronize(sc->stream);
  {
    std::lock_guard<std::mutex> lk(pool_mu_);
    sc->in_use = false;
  }
  pool_cv_.notify_one();
}

int Engine::search(int nq, const float *xq, int k, int nprobe,
                   int recall_num, int metric, bool brute_force,
                   const std::string &request_id, int partition_id,
                   float *out_dists, int64_t *out_ids, bool l2_sqrt,
                   const std::vector<TermFilterSpec> *term_filters,
                   const std::vector<RangeFilterSpec> *range_filters,
                   std::string *filter_err, int filter_op, bool prelocked) {
  if (!table_created_ || nq <= 0 || k <= 0) return -1;
  /* Concurrent searches: each takes the read lock (Add/Build/Load are
   * write-locked) plus one SearchScratch context from the pool — its
   * own HIP stream and device buffers — so arbitrary cgo threads can
   * search while a background thread indexes (engine.cc:1108-1127,
   * SURVEY 8b). The ScratchGuard synchronizes the context's stream
   * before release, so no async work survives the read lock. The C ABI
   * passes prelocked=true when it already holds read_lock() across
   * search + response assembly. */
  std::shared_lock<std::shared_mutex> g(rw_, std::defer_lock);
  if (!prelocked) g.lock();
  SearchScratch *scp = acquire_scratch_();
  if (!scp) return -1;
  ScratchGuard sg{this, scp};
  SearchScratch &sc = *scp;
  const int pid = partition_id;
  bool ip = metric == 0 ? params_.metric_ip : (metric == 2);
  hipStream_t s = sc.stream;

  struct Timer {
    hipEvent_t ev[6];
    hipStream_t s;
    explicit Timer(hipStream_t s_) : s(s_) {
      for (auto &e : ev) (void)hipEventCreate(&e);
    }
    ~Timer() {
      for (auto &e : ev) (void)hipEventDestroy(e);
    }
    void rec(int i) { (void)hipEventRecord(ev[i], s); }
    double ms(int a, int b) {
      float m = 0;
      (void)hipEventElapsedTime(&m, ev[a], ev[b]);
      return m;
    }
  } tm(s);

  if (KillRegistry::inst().killed(request_id, pid)) return -2;

  /* k2: scan-phase candidates; recall_num>k -> rerank leg
   * (search_preassigned, ivfpq.cc:765-776) */
  int k2 = std::max(k, recall_num);
  bool rerank = recall_num > 0;
  if (k2 > 1024) {
    if (filter_err)
      *filter_err = "topN+offset (or recall_num) = " + std::to_string(k2) +
                    " exceeds this engine's supported maximum of 1024";
    return -3;
  }

  tm.rec(0);
  const float *qptr = nullptr;
  if (xq != nullptr) {
    if (sc.q_dev.reserve((size_t)nq * dim_ * 4)) return -1;
    GAMMA_CHECK(hipMemcpyAsync(sc.q_dev.get(), xq, (size_t)nq * dim_ * 4,
                               hipMemcpyHostToDevice, s));
    qptr = sc.q_dev.as<float>();
  } else if (nq != cached_nq_) {
    return -1; /* cache_queries() first */
  } else {
    qptr = cached_q_dev_.as<float>();
  }
  if (sc.q_norms.reserve((size_t)nq * 4)) return -1;
  GAMMA_CHECK(gk::row_norms(s, qptr, nq, dim_, sc.q_norms.as<float>()));
  if (sc.keys.reserve((size_t)nq * k2 * 8)) return -1;
  if (sc.out_d.reserve((size_t)nq * k * 4)) return -1;
  if (sc.out_i.reserve((size_t)nq * k * 8)) return -1;
  tm.rec(1);

  const uint32_t *bm = bitmap_.any() ? bitmap_.dev() : nullptr;
  bool have_filters = (term_filters && !term_filters->empty()) ||
                      (range_filters && !range_filters->empty());
  if (have_filters) {
    static const std::vector<TermFilterSpec> kNoT;
    static const std::vector<RangeFilterSpec> kNoR;
    if (build_filter_bitmap_(term_filters ? *term_filters : kNoT,
                             range_filters ? *range_filters : kNoR, sc,
                             &bm, filter_err, filter_op))
      return -3;
  }

  bool use_flat = brute_force || params_.kind == IndexKind::FLAT ||
                  !index_ || !index_->trained();
  double t_assign = 0, t_scan = 0;
  bool need_canonical_rerank = true; /* FLAT & IVFFLAT canonicalize */
  if (use_flat) {
    /* FLAT margin so GEMM-order rounding cannot evict a true top-k
     * member before the canonical re-rank (DESIGN.md numerics note) */
    int kf = std::min<int64_t>((int64_t)k2 + 64, 1088);
    kf = (int)std::min<int64_t>(kf, std::max<int64_t>(raw_.size(), 1));
    if (sc.keys.reserve((size_t)nq * kf * 8)) return -1;
    if (flat_search_keys(raw_, dim_, qptr, nq, kf,
                         sc.q_norms.as<float>(), ip, s,
                         sc.keys.as<uint64_t>(), bm, sc))
      return -1;
    k2 = kf;
  } else {
    /* nprobe unset -> the index's configured nprobe (the reference's
     * retrieval-params default, ivfpq.cc:253-294 / ivfpq.h:1049) */
    if (nprobe <= 0) nprobe = index_->params().nprobe;
    if (nprobe <= 0) nprobe = 80;
    int S = index_->probe_split(nq, k2, nprobe);
    if (S > 1 && sc.keys.reserve((size_t)nq * S * k2 * 8)) return -1;
    /* arm the in-flight kill flag so SetKillStatus stops the scan
     * between lists (request_context.h:83 semantics) */
    const int *kf = nullptr;
    if (!request_id.empty()) {
      if (sc.kill_flag.reserve(4)) return -1;
      GAMMA_CHECK(hipMemsetAsync(sc.kill_flag.get(), 0, 4, s));
      KillRegistry::inst().arm(request_id, pid, sc.kill_flag.as<int>());
      kf = sc.kill_flag.as<int>();
    }
    int rc_idx = index_->search(qptr, nq, k2, nprobe, bm, ip, s,
                                sc.keys.as<uint64_t>(),
                                sc.q_norms.as<float>(), &t_assign,
                                &t_scan, sc, S, kf);
    if (kf) KillRegistry::inst().disarm(sc.kill_flag.as<int>());
    if (rc_idx) return -1;
    k2 = k2 * S; /* sub-block partials merge in the sort below */
    /* ADC distances already match the oracle bit-for-bit; canonical
     * re-rank only when the caller asked for the exact rerank leg */
    /* (sub-block partials are unsorted relative to each other, but
     * sort_rows below always sorts — no extra pass needed for S>1) */
    need_canonical_rerank =
        rerank || params_.kind == IndexKind::IVFFLAT;
  }
  tm.rec(2);
  if (KillRegistry::inst().killed(request_id, pid)) return -2;

  if (need_canonical_rerank) {
    GAMMA_CHECK(gk::rerank(s, nq, k2, dim_, qptr, raw_.dev_seg_table(),
                           raw_.num_segs(), RawStore::SEG_SHIFT, ip,
                           sc.keys.as<uint64_t>(),
                           sc.keys.as<uint64_t>()));
  }
  GAMMA_CHECK(gk::sort_rows(s, nq, k2, k, sc.keys.as<uint64_t>(), ip,
                            sc.out_d.as<float>(),
                            sc.out_i.as<int64_t>()));
  tm.rec(3);
  GAMMA_CHECK(hipMemcpyAsync(out_dists, sc.out_d.get(),
                             (size_t)nq * k * 4, hipMemcpyDeviceToHost, s));
  GAMMA_CHECK(hipMemcpyAsync(out_ids, sc.out_i.get(), (size_t)nq * k * 8,
                             hipMemcpyDeviceToHost, s));
  tm.rec(4);
  GAMMA_CHECK(hipStreamSynchronize(s));

  if (l2_sqrt && !ip) {
    for (int64_t i = 0; i < (int64_t)nq * k; i++)
      if (out_ids[i] >= 0) out_dists[i] = sqrtf(out_dists[i]);
  }
  {
    std::lock_guard<std::mutex> lk(timing_mu_);
    last_timing[0] = tm.ms(0, 1) * 1000.0;
    last_timing[1] = (use_flat ? 0.0 : t_assign) * 1000.0;
    last_timing[2] = (use_flat ? tm.ms(1, 2) : t_scan) * 1000.0;
    last_timing[3] = tm.ms(2, 3) * 1000.0;
    last_timing[4] = tm.ms(3, 4) * 1000.0;
    last_timing[5] = tm.ms(0, 4) * 1000.0;
  }
  return 0;
}

int Engine::search_field_(RawStore &raw, IVFIndex *idx, int dim,
                          const float *xq, int nq, int topn, int nprobe,
                          int recall_num, bool ip, bool brute_force,
                          SearchScratch &sc, const uint32_t *bm,
                          float *host_dists, int64_t *host_ids) {
  hipStream_t s = sc.stream;
  int k2 = std::max(topn, recall_num);
  bool rr = recall_num > 0;
  if (k2 > 1024) return -1;
  if (sc.q_dev.reserve((size_t)nq * dim * 4)) return -1;
  GAMMA_CHECK(hipMemcpyAsync(sc.q_dev.get(), xq, (size_t)nq * dim * 4,
                             hipMemcpyHostToDevice, s));
  const float *qptr = sc.q_dev.as<float>();
  if (sc.q_norms.reserve((size_t)nq * 4)) return -1;
  GAMMA_CHECK(gk::row_norms(s, qptr, nq, dim, sc.q_norms.as<float>()));
  if (sc.out_d.reserve((size_t)nq * topn * 4)) return -1;
  if (sc.out_i.reserve((size_t)nq * topn * 8)) return -1;
  bool use_flat = brute_force || !idx || !idx->trained();
  bool canon = true;
  if (use_flat) {
    int kf = std::min<int64_t>((int64_t)k2 + 64, 1088);
    kf = (int)std::min<int64_t>(kf, std::max<int64_t>(raw.size(), 1));
    if (sc.keys.reserve((size_t)nq * kf * 8)) return -1;
    if (flat_search_keys(raw, dim, qptr, nq, kf, sc.q_norms.as<float>(),
                         ip, s, sc.keys.as<uint64_t>(), bm, sc))
      return -1;
    k2 = kf;
  } else {
    if (nprobe <= 0) nprobe = idx->params().nprobe;
    if (nprobe <= 0) nprobe = 80;
    int S = idx->probe_split(nq, k2, nprobe);
    if (sc.keys.reserve((size_t)nq * S * k2 * 8)) return -1;
    double ta = 0, ts = 0;
    if (idx->search(qptr, nq, k2, nprobe, bm, ip, s,
                    sc.keys.as<uint64_t>(), sc.q_norms.as<float>(), &ta,
                    &ts, sc, S, nullptr))
      return -1;
    k2 *= S;
    canon = rr || params_.kind == IndexKind::IVFFLAT;
  }
  if (canon)
    GAMMA_CHECK(gk::rerank(s, nq, k2, dim, qptr, raw.dev_seg_table(),
                           raw.num_segs(), RawStore::SEG_SHIFT, ip,
                           sc.keys.as<uint64_t>(),
                           sc.keys.as<uint64_t>()));
  GAMMA_CHECK(gk::sort_rows(s, nq, k2, topn, sc.keys.as<uint64_t>(), ip,
                            sc.out_d.as<float>(), sc.out_i.as<int64_t>()));
  GAMMA_CHECK(hipMemcpyAsync(host_dists, sc.out_d.get(),
                             (size_t)nq * topn * 4, hipMemcpyDeviceToHost,
                             s));
  GAMMA_CHECK(hipMemcpyAsync(host_ids, sc.out_i.get(),
                             (size_t)nq * topn * 8, hipMemcpyDeviceToHost,
                             s));
  GAMMA_CHECK(hipStreamSynchronize(s));
  return 0;
}

int Engine::search_multi(int nq, const std::vector<MultiVecQuery> &queries,
                         int topn, int nprobe, int recall_num, int metric,
                         bool brute_force, const std::string &request_id,
                         int partition_id,
                         const std::vector<double> &weights,
                         bool rank_by_score, double *out_scores,
                         int64_t *out_ids, std::string *err,
                         const std::vector<TermFilterSpec> *term_filters,
                         const std::vector<RangeFilterSpec> *range_filters,
                         int filter_op, bool prelocked) {
  if (!table_created_ || nq <= 0 || topn <= 0) return -1;
  const size_t vn = queries.size();
  if (vn < 2) {
    if (err) *err = "search_multi needs >= 2 vector queries";
    return -1;
  }
  bool ip = metric == 0 ? params_.metric_ip : (metric == 2);
  struct FieldRef {
    RawStore *raw;
    IVFIndex *idx;
    int dim;
  };
  std::vector<FieldRef> fs;
  for (auto &q : queries) {
    if (q.name == vec_name_) {
      fs.push_back({&raw_, index_.get(), dim_});
    } else if (ExtraVecField *e = extra_vec_(q.name)) {
      fs.push_back({&e->raw, e->index.get(), e->dim});
    } else {
      if (err) *err = "unknown vector field " + q.name;
      return -1;
    }
  }
  std::shared_lock<std::shared_mutex> g(rw_, std::defer_lock);
  if (!prelocked) g.lock();
  if (KillRegistry::inst().killed(request_id, partition_id)) return -2;
  SearchScratch *scp = acquire_scratch_();
  if (!scp) return -1;
  ScratchGuard sg{this, scp};
  /* docid-level filters apply to every field's scan identically */
  const uint32_t *bm = bitmap_.any() ? bitmap_.dev() : nullptr;
  bool have_filters = (term_filters && !term_filters->empty()) ||
                      (range_filters && !range_filters->empty());
  if (have_filters) {
    static const std::vector<TermFilterSpec> kNoT;
    static const std::vector<RangeFilterSpec> kNoR;
    if (build_filter_bitmap_(term_filters ? *term_filters : kNoT,
                             range_filters ? *range_filters : kNoR, *scp,
                             &bm, err, filter_op))
      return -1;
  }
  /* per field: the field's own index to depth topn
   * (vector_manager.cc:955: topN = condition->topn when merging) */
  std::vector<std::vector<float>> fd(vn);
  std::vector<std::vector<int64_t>> fi(vn);
  for (size_t j = 0; j < vn; j++) {
    fd[j].resize((size_t)nq * topn);
    fi[j].resize((size_t)nq * topn);
    if (search_field_(*fs[j].raw, fs[j].idx, fs[j].dim, queries[j].vecs,
                      nq, topn, nprobe, recall_num, ip, brute_force,
                      *scp, bm, fd[j].data(), fi[j].data()))
      return -1;
    if (KillRegistry::inst().killed(request_id, partition_id)) return -2;
  }
  /* merge (vector_manager.cc:1025-1086): a doc survives iff it is in
   * EVERY field's top-n; score = sum_j weight_j * dist_j
   * (WeightedRanker, default 1/vec_num); docid-ascending order, or
   * combined-score order when multi_vector_rank is set */
  for (int i = 0; i < nq; i++) {
    std::vector<std::vector<std::pair<int64_t, float>>> lists(vn);
    for (size_t j = 0; j < vn; j++) {
      const MultiVecQuery &mq = queries[j];
      for (int t = 0; t < topn; t++) {
        int64_t id = fi[j][(size_t)i * topn + t];
        if (id < 0) continue;
        float dj = fd[j][(size_t)i * topn + t];
        /* per-field score range (IsSimilarScoreValid) */
        if (mq.has_min && dj < mq.min_score) continue;
        if (mq.has_max && dj > mq.max_score) continue;
        lists[j].push_back({id, dj});
      }
      std::sort(lists[j].begin(), lists[j].end());
    }
    std::vector<std::pair<double, int64_t>> merged; /* (score, docid) */
    for (auto &pr : lists[0]) {
      double score = 0;
      bool common = true;
      for (size_t j = 0; j < vn && common; j++) {
        auto it = std::lower_bound(
            lists[j].begin(), lists[j].end(), pr.first,
            [](const std::pair<int64_t, float> &a, int64_t b) {
              return a.first < b;
            });
        if (it == lists[j].end() || it->first != pr.first) {
          common = false;
        } else {
          double w = weights.size() == vn ? weights[j] : 1.0 / vn;
          score += it->second * w;
        }
      }
      if (common) merged.push_back({score, pr.first});
    }
    if (rank_by_score) {
      if (ip) /* InnerProductCmp: best (largest) first */
        std::stable_sort(merged.begin(), merged.end(),
                         [](const std::pair<double, int64_t> &a,
                            const std::pair<double, int64_t> &b) {
                           return a.first > b.first;
                         });
      else /* L2Cmp: smallest first */
        std::stable_sort(merged.begin(), merged.end());
    } /* else: docid-ascending (lists[0] order is already sorted) */
    for (int t = 0; t < topn; t++) {
      if (t < (int)merged.size()) {
        out_scores[(size_t)i * topn + t] = merged[t].first;
        out_ids[(size_t)i * topn + t] = merged[t].second;
      } else {
        out_scores[(size_t)i * topn + t] = 0;
        out_ids[(size_t)i * topn + t] = -1;
      }
    }
  }
  return 0;
}

int Engine::filter_docids(const std::vector<TermFilterSpec> &terms,
                          const std::vector<RangeFilterSpec> &ranges,
                          int offset, int limit, std::vector<int64_t> *out,
                          std::string *err, int filter_op, bool prelocked) {
  std::shared_lock<std::shared_mutex> g(rw_, std::defer_lock);
  if (!prelocked) g.lock();
  SearchScratch *scp = acquire_scratch_();
  if (!scp) return -1;
  ScratchGuard sg{this, scp};
  if (build_filter_bitmap_(terms, ranges, *scp, nullptr, err, filter_op))
    return -1;
  int64_t skipped = 0;
  for (int64_t id = 0; id < max_docid_; id++) {
    if ((scp->filt_host[id >> 5] >> (id & 31)) & 1u) continue;
    if (skipped++ < offset) continue;
    out->push_back(id);
    if (limit > 0 && (int)out->size() >= limit) break;
  }
  return 0;
}

int Engine::cache_queries(int nq, const float *xq) {
  std::unique_lock<std::shared_mutex> g(rw_);
  if (cached_q_dev_.reserve((size_t)nq * dim_ * 4)) return -1;
  GAMMA_CHECK(hipMemcpy(cached_q_dev_.get(), xq, (size_t)nq * dim_ * 4,
                        hipMemcpyHostToDevice));
  cached_nq_ = nq;
  return 0;
}

int64_t Engine::docid_of(const std::string &p_key) const {
  std::lock_guard<std::mutex> pk(pkey_mu_);
  auto it = pkey2docid_.find(p_key);
  return it == pkey2docid_.end() ? -1 : it->second;
}

const std::string &Engine::pkey_of(int64_t docid) const {
  if (docid < 0 || docid >= (int64_t)docid2pkey_.size()) return kEmpty;
  return docid2pkey_[docid];
}

const std::string *Engine::field_value(int64_t docid,
                                       const std::string &f) const {
  auto it = field_vals_.find(f);
  if (it == field_vals_.end()) return nullptr;
  if (docid < 0 || docid >= (int64_t)it->second.size()) return nullptr;
  return &it->second[docid];
}

std::string Engine::status_json() const {
  char buf[512];
  snprintf(buf, sizeof buf,
           "{\"doc_count\": %lld, \"index_status\": %d, \"min_indexed_num\": "
           "%lld, \"max_docid\": %lld, \"table_name\": \"%s\"}",
           (long long)(max_docid_.load() - bitmap_.popcount()),
           (index_ && index_->trained()) ? 2 : 0,
           (long long)indexed_count_.load(),
           (long long)(max_docid_.load() - 1),
           gjson::escape(space_name_).c_str());
  return buf;
}

static const uint32_t kDumpMagic = 0x47414D41; /* "GAMA" */

int Engine::dump(std::string *err) { return dump_to_(path_, err); }

int Engine::dump_to_(const std::string &dir, std::string *err) {
  std::unique_lock<std::shared_mutex> g(rw_);
  mkdir(dir.c_str(), 0755);
  std::string fn = dir + "/gamma.dump";
  FILE *f = fopen((fn + ".tmp").c_str(), "wb");
  if (!f) {
    if (err) *err = "cannot open dump file " + fn;
    return -1;
  }
  fwrite(&kDumpMagic, 4, 1, f);
  int ver = 2; /* v2 appends the extra vector fields after the index */
  fwrite(&ver, 4, 1, f);
  auto wstr = [&](const std::string &s_) {
    int64_t n = (int64_t)s_.size();
    fwrite(&n, 8, 1, f);
    fwrite(s_.data(), 1, n, f);
  };
  wstr(space_name_);
  wstr(index_type_);
  wstr(vec_name_);
  fwrite(&dim_, 4, 1, f);
  fwrite(&training_threshold_, 4, 1, f);
  int64_t maxdoc_now = max_docid_.load(std::memory_order_acquire);
  fwrite(&maxdoc_now, 8, 1, f);
  int nfields = (int)fields_.size();
  fwrite(&nfields, 4, 1, f);
  for (auto &fm : fields_) {
    wstr(fm.name);
    fwrite(&fm.data_type, 4, 1, f);
  }
  for (int64_t i = 0; i < maxdoc_now; i++) wstr(docid2pkey_[i]);
  for (auto &fm : fields_) {
    auto &col = field_vals_.at(fm.name);
    for (int64_t i = 0; i < maxdoc_now; i++)
      wstr(i < (int64_t)col.size() ? col[i] : kEmpty);
  }
  raw_.dump(f);
  bitmap_.dump(f);
  int has_index = index_ ? 1 : 0;
  fwrite(&has_index, 4, 1, f);
  if (index_) index_->dump(f, stream_);
  /* v2: extra vector fields */
  int n_extra = (int)extra_vecs_.size();
  fwrite(&n_extra, 4, 1, f);
  for (auto &e : extra_vecs_) {
    wstr(e->name);
    fwrite(&e->dim, 4, 1, f);
    e->raw.dump(f);
    int hi = (e->index && e->index->trained()) ? 1 : 0;
    fwrite(&hi, 4, 1, f);
    if (hi) e->index->dump(f, stream_);
  }
  fclose(f);
  rename((fn + ".tmp").c_str(), fn.c_str());
  return 0;
}

int Engine::load(std::string *err) {
  std::unique_lock<std::shared_mutex> g(rw_);
  std::string fn = path_ + "/gamma.dump";
  FILE *f = fopen(fn.c_str(), "rb");
  if (!f) {
    /* nothing dumped yet: reference Load returns ok on empty dir */
    return 0;
  }
  uint32_t magic = 0;
  int ver = 0;
  if (fread(&magic, 4, 1, f) != 1 || magic != kDumpMagic ||
      fread(&ver, 4, 1, f) != 1) {
    fclose(f);
    if (err) *err = "bad dump file";
    return -1;
  }
  auto rstr = [&](std::string &s_) {
    int64_t n = 0;
    if (fread(&n, 8, 1, f) != 1 || n < 0 || n > (int64_t)1 << 30) return -1;
    s_.resize(n);
    if (n && fread(&s_[0], 1, n, f) != (size_t)n) return -1;
    return 0;
  };
  std::string idx_type, vecn;
  if (rstr(space_name_) || rstr(idx_type) || rstr(vecn)) {
    fclose(f);
    return -1;
  }
  int dim = 0, tt = 0;
  int64_t maxdoc = 0;
  auto bad = [&](const char *what) {
    fclose(f);
    if (err) *err = std::string("bad dump file: ") + what;
    return -1;
  };
  if (fread(&dim, 4, 1, f) != 1 || fread(&tt, 4, 1, f) != 1 ||
      fread(&maxdoc, 8, 1, f) != 1)
    return bad("truncated header");
  if (dim <= 0 || dim > (1 << 20) || maxdoc < 0 ||
      maxdoc > ((int64_t)1 << 40))
    return bad("implausible dim/maxdoc");
  if (table_created_ && dim != dim_) {
    fclose(f);
    if (err)
      *err = "dump dimension " + std::to_string(dim) +
             " != table dimension " + std::to_string(dim_);
    return -1;
  }
  if (table_created_ && idx_type != index_type_) {
    fclose(f);
    if (err)
      *err = "dump index type " + idx_type + " != table " + index_type_;
    return -1;
  }
  int nfields = 0;
  if (fread(&nfields, 4, 1, f) != 1 || nfields < 0 || nfields > 65536)
    return bad("field count");
  std::vector<FieldMeta> fms(nfields);
  for (auto &fm : fms) {
    if (rstr(fm.name)) return bad("field name");
    if (fread(&fm.data_type, 4, 1, f) != 1) return bad("field type");
  }
  if (!table_created_) {
    /* table should have been created before Load (reference flow);
     * recreate from the dump if not */
    std::string e2;
    if (create_table(space_name_, fms, vecn, dim, idx_type, "", tt, &e2)) {
      fclose(f);
      if (err) *err = e2;
      return -1;
    }
  }
  max_docid_.store(maxdoc, std::memory_order_release);
  docid2pkey_.resize(maxdoc);
  {
    std::lock_guard<std::mutex> pk(pkey_mu_);
    for (int64_t i = 0; i < maxdoc; i++) {
      if (rstr(docid2pkey_[i])) { fclose(f); return -1; }
      pkey2docid_[docid2pkey_[i]] = i;
    }
  }
  for (auto &fm : fms) {
    auto &col = field_vals_[fm.name];
    col.resize(maxdoc);
    for (int64_t i = 0; i < maxdoc; i++)
      if (rstr(col[i])) { fclose(f); return -1; }
  }
  if (raw_.load(f, stream_)) { fclose(f); return -1; }
  if (bitmap_.load(f, stream_)) { fclose(f); return -1; }
  bitmap_.ensure(maxdoc, stream_);
  int has_index = 0;
  if (fread(&has_index, 4, 1, f) != 1) return bad("index flag");
  if (has_index && index_) {
    if (index_->load(f, stream_)) { fclose(f); return -1; }
    indexed_count_ = index_->ntotal();
  }
  if (ver >= 2) {
    int n_extra = 0;
    if (fread(&n_extra, 4, 1, f) != 1 || n_extra < 0 || n_extra > 64)
      return bad("extra vec count");
    if (n_extra != (int)extra_vecs_.size())
      return bad("extra vector field count mismatch with table");
    for (int i = 0; i < n_extra; i++) {
      std::string ename;
      int edim = 0;
      if (rstr(ename)) return bad("extra vec name");
      if (fread(&edim, 4, 1, f) != 1) return bad("extra vec dim");
      ExtraVecField *e = extra_vec_(ename);
      if (!e || e->dim != edim)
        return bad("extra vector field mismatch with table");
      if (e->raw.load(f, stream_)) return bad("extra raw vectors");
      int hi = 0;
      if (fread(&hi, 4, 1, f) != 1) return bad("extra index flag");
      if (hi && e->index && e->index->load(f, stream_))
        return bad("extra index");
    }
  } else if (!extra_vecs_.empty()) {
    return bad("v1 dump has no extra vector fields");
  }
  fclose(f);
  return 0;
}

int Engine::add_field_index(const std::string &name,
                            const std::vector<std::string> &fields,
                            const std::string &index_type,
                            const std::string &params, std::string *err) {
  (void)index_type;
  (void)params;
  if (!table_created_) {
    if (err) *err = "table not initialized";
    return -1;
  }
  if (name.empty()) {
    if (err) *err = "index_name is empty";
    return -1;
  }
  if (fields.empty()) {
    if (err) *err = "field_names is empty";
    return -1;
  }
  std::unique_lock<std::shared_mutex> g(rw_);
  if (named_indexes_.count(name)) return 0; /* exists: ignore
                                               (engine.cc:1578) */
  if (fields[0] == vec_name_) {
    /* vector index: exactly one field; the vector index itself is
     * built by BuildIndex — this registers the name (engine.cc:1593) */
    if (fields.size() != 1) {
      if (err) *err = "vector index must reference exactly one field";
      return -1;
    }
  } else {
    /* scalar path: validate every field synchronously (engine.cc:1611),
     * then build its ScalarFieldIndex eagerly so the first filtered
     * search pays nothing */
    for (auto &fname : fields) {
      int dt = -1;
      for (auto &fm : fields_)
        if (fm.name == fname) dt = fm.data_type;
      if (dt < 0) {
        if (err) *err = "field [" + fname + "] not found in table";
        return -1;
      }
      std::lock_guard<std::mutex> lk(scalar_mu_);
      extend_scalar_index_(fname, dt, true, true, max_docid_);
    }
  }
  named_indexes_[name] = fields;
  return 0;
}

int Engine::remove_field_index(const std::string &name, std::string *err) {
  if (!table_created_) {
    if (err) *err = "table not initialized";
    return -1;
  }
  if (name.empty()) {
    if (err) *err = "index_name is empty";
    return -1;
  }
  std::unique_lock<std::shared_mutex> g(rw_);
  auto it = named_indexes_.find(name);
  if (it == named_indexes_.end()) return 0; /* idempotent
                                               (scalar_index_manager.h
                                               RemoveIndex) */
  for (auto &fname : it->second) {
    if (fname == vec_name_) continue; /* the vector index itself stays;
                                         only the name is dropped */
    std::lock_guard<std::mutex> lk(scalar_mu_);
    scalar_idx_.erase(fname); /* filtered search rebuilds lazily */
  }
  named_indexes_.erase(it);
  return 0;
}

int Engine::set_config(const std::string &json) {
  gjson::Value v;
  if (!gjson::parse(json, v)) return -1;
  /* engine_cache_size / enable_id_cache: RocksDB-side knobs, accepted
   * and ignored on the GPU engine (engine.cc:2087-2112 fields) */
  std::string p;
  if (v.get_str("path", p) && !p.empty()) path_ = p;
  int x;
  if (v.get_int("slow_search_time", x)) slow_search_time_ = x;
  if (v.get_int("refresh_interval", x)) refresh_interval_ = x;
  return 0;
}

std::string Engine::get_config() const {
  char buf[512];
  snprintf(buf, sizeof buf,
           "{\"engine_cache_size\": 0, \"path\": \"%s\", "
           "\"slow_search_time\": %d, \"refresh_interval\": %d, "
           "\"enable_id_cache\": true}",
           gjson::escape(path_).c_str(), slow_search_time_,
           refresh_interval_);
  return buf;
}

int Engine::backup(int command, std::string *err) {
  if (command == 0) { /* create (BackupThread command 0) */
    mkdir(path_.c_str(), 0755); /* parent first (utils::make_dir) */
    return dump_to_(path_ + "/backup", err);
  }
  /* other commands: the reference's BackupThread does nothing and
   * reports success (engine.cc:1529-1553) */
  return 0;
}

}  // namespace vgamma
