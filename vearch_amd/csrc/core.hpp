/*
 * core.hpp — MI355X-native Gamma engine core: table-lite, raw-vector
 * store, delete bitmap, realtime inverted lists, FLAT/IVFFLAT/IVFPQ
 * index models, train/encode, dump/load, kill registry.
 *
 * Thin reimplementation of the reference layers below the C ABI
 * (search/engine.cc, vector/vector_manager.cc, index/impl/(star),
 * index/realtime, vector/memory_raw_vector) with the compute on the
 * GPU (kernels.hip). GPU state is the product path; the host shadow of
 * raw vectors exists for training and Dump only.
 */
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#include <atomic>
#include <map>
#include <memory>
#include <mutex>
#include <shared_mutex>
#include <condition_variable>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "kernels.h"

namespace vgamma {

#define GAMMA_CHECK(x)                                            \
  do {                                                            \
    hipError_t err__ = (x);                                       \
    if (err__ != hipSuccess) {                                    \
      fprintf(stderr, "[gamma] HIP error %s at %s:%d: %s\n",      \
              hipGetErrorString(err__), __FILE__, __LINE__, #x);  \
      return -1;                                                  \
    }                                                             \
  } while (0)

/* kill registry — RequestContext analog (c_api/api_data/request_context.h:
 * 51-99): SetKillStatus marks (request_id, partition_id); in-flight
 * searches poll it between kernel stages and return -2. */
class KillRegistry {
 public:
  static KillRegistry &inst();
  void set(const std::string &rid, int pid);
  void del(const std::string &rid, int pid);
  bool killed(const std::string &rid, int pid);
  /* in-flight searches register a device int; SetKillStatus writes 1 to
   * it so running scan kernels stop between lists (ivfpq.h:927
   * is_killed_every analog) */
  void arm(const std::string &rid, int pid, int *dev_flag);
  void disarm(int *dev_flag);

 private:
  std::mutex mu_;
  std::map<std::pair<std::string, int>, int> map_;
  std::map<int *, std::pair<std::string, int>> armed_;
};

class DeviceBuf {
 public:
  DeviceBuf() = default;
  ~DeviceBuf() { free(); }
  DeviceBuf(const DeviceBuf &) = delete;
  DeviceBuf &operator=(const DeviceBuf &) = delete;
  int reserve(size_t bytes);         /* grow-only */
  void free();
  /* take other's allocation (frees our own); other becomes empty */
  void steal(DeviceBuf &other);
  void *get() const { return p_; }
  template <class T> T *as() const { return (T *)p_; }
  size_t bytes() const { return bytes_; }

 private:
  void *p_ = nullptr;
  size_t bytes_ = 0;
};

/* Append-only string column with stable element storage: readers under
 * the engine's shared lock may index any element below the published
 * doc count while an appender (also under the shared lock — the
 * lock-free add path, realtime_mem_data.cc:57-68 publication order)
 * pushes new rows. Blocks never move once allocated and the block
 * table's storage is reserved up front, so existing elements are never
 * relocated by growth. */
class StableStrCol {
  static constexpr int SHIFT = 13; /* 8192 strings per block */
  static constexpr size_t BMASK = ((size_t)1 << SHIFT) - 1;
  using Block = std::vector<std::string>;

 public:
  StableStrCol() { blocks_.reserve((size_t)1 << (31 - SHIFT)); }
  size_t size() const { return size_.load(std::memory_order_acquire); }
  void push_back(std::string v) {
    size_t i = size_.load(std::memory_order_relaxed);
    size_t b = i >> SHIFT;
    if (b == blocks_.size()) {
      blocks_.push_back(std::make_unique<Block>());
      blocks_.back()->resize((size_t)1 << SHIFT);
    }
    (*blocks_[b])[i & BMASK] = std::move(v);
    size_.store(i + 1, std::memory_order_release);
  }
  void resize(size_t n) { /* grow-only */
    while (size_.load(std::memory_order_relaxed) < n)
      push_back(std::string());
  }
  /* idempotent write-at-index (an aborted append may retry the same
   * docid; a blind push_back would shift every later row) */
  void set(size_t i, std::string v) {
    resize(i + 1);
    (*blocks_[i >> SHIFT])[i & BMASK] = std::move(v);
  }
  std::string &operator[](size_t i) { return (*blocks_[i >> SHIFT])[i & BMASK]; }
  const std::string &operator[](size_t i) const {
    return (*blocks_[i >> SHIFT])[i & BMASK];
  }

 private:
  std::vector<std::unique_ptr<Block>> blocks_;
  std::atomic<size_t> size_{0};
};

/* Raw vectors: fp32 row-major, append-only segments of 2^SEG_SHIFT
 * vectors (memory_raw_vector.h:58 segment array). Device segments are
 * the product store; host shadow serves training + Dump. */
class RawStore {
 public:
  static constexpr int SEG_SHIFT = 19; /* 512k vectors per segment */
  int init(int d);
  int add(const float *x, int64_t cnt, hipStream_t s); /* host pointer */
  /* true if add(cnt) would allocate (new segment / norms growth /
   * seg-table upload) — the lock-free append path requires false */
  bool would_grow(int64_t cnt) const {
    int64_t n_new = n_.load(std::memory_order_relaxed) + cnt;
    return (size_t)((n_new + ((int64_t)1 << SEG_SHIFT) - 1) >> SEG_SHIFT) >
               dev_segs_.size() ||
           n_new > norms_cap_;
  }
  int64_t size() const { return n_.load(std::memory_order_acquire); }
  int dim() const { return d_; }
  const float *host_row(int64_t vid) const;
  /* copy [start,start+cnt) rows into a contiguous host buffer */
  void host_copy(int64_t start, int64_t cnt, float *out) const;
  const float *const *dev_seg_table() const {
    return (const float *const *)seg_table_.get();
  }
  int num_segs() const { return (int)dev_segs_.size(); }
  const float *dev_norms() const { return norms_.as<float>(); }
  /* contiguous device run beginning at vid (within one segment) */
  const float *dev_run(int64_t vid, int64_t *run_len) const;
  int dump(FILE *f) const;
  int load(FILE *f, hipStream_t s);

 private:
  int ensure_capacity(int64_t n_new, hipStream_t s);
  int d_ = 0;
  /* published AFTER data + norms are resident (release), so a reader
   * holding only the shared lock sees a consistent prefix */
  std::atomic<int64_t> n_{0};
  std::vector<void *> dev_segs_;
  std::vector<std::vector<float>> host_segs_;
  DeviceBuf seg_table_;
  DeviceBuf norms_;
  int64_t norms_cap_ = 0;
};

/* docid delete bitmap (util/bitmap_manager analog) with device mirror */
class Bitmap {
 public:
  int set(int64_t vid, hipStream_t s); /* mark deleted */
  bool test(int64_t vid) const;
  bool any() const { return set_count_ > 0; }
  int ensure(int64_t nbits, hipStream_t s);
  bool has_capacity(int64_t nbits) const { return nbits <= bits_; }
  const uint32_t *dev() const { return dev_.as<uint32_t>(); }
  /* host shadow word (0 when out of range) for filter-bitmap merges */
  uint32_t host_word(int64_t w) const {
    return (w >= 0 && w < (int64_t)host_.size()) ? host_[w] : 0;
  }
  int64_t popcount() const;
  int dump(FILE *f) const;
  int load(FILE *f, hipStream_t s);

 private:
  std::vector<uint32_t> host_;
  DeviceBuf dev_;
  int64_t bits_ = 0;
  int64_t set_count_ = 0;
};

enum class IndexKind { FLAT, IVFPQ, IVFFLAT };

struct IndexParams {
  IndexKind kind = IndexKind::IVFPQ;
  int ncentroids = 2048;       /* ivfpq.h:1049 default */
  int nsubvector = 0;          /* 0 -> d/2 (ivfpq.cc:122-124) */
  int nbits = 8;
  int nprobe = 80;
  int bucket_init_size = 1000;
  int bucket_max_size = 1280000;
  int training_threshold = 0;
  bool metric_ip = true;       /* reference default INNER_PRODUCT */
  /* OPQ pre-rotation (ivfpq.h:1043-1044 has_opq/opq_nsubvector;
   * OPQMatrix(d, opq_nsubvector, d) at ivfpq.cc:177) */
  bool has_opq = false;
  int opq_nsubvector = 0;
  /* parse the index-params JSON (ivfpq.h:1065 Parse); empty ok */
  int parse(const std::string &json, std::string *err);
};

/* Realtime inverted lists + trained model, device-resident.
 * Semantics follow index/realtime/realtime_mem_data.cc: per-bucket SoA
 * (int64 ids with bit-63 delete mask, packed codes), append-only with
 * capacity extension; size published after the data copy so concurrent
 * scans see a consistent prefix. */
/* Per-search scratch context: one HIP stream + every device buffer a
 * search touches. A pool of these (Engine::kMaxConcurrentSearches)
 * makes concurrent cgo Search calls truly concurrent (engine.cc
 * 1108-1127 threading contract): searches share the engine
 * read-locked, each on its own stream; Add/Build/Load take the write
 * lock. Grow-only buffers, so steady-state searches never allocate. */
struct SearchScratch {
  hipStream_t stream = nullptr;
  /* engine-side */
  DeviceBuf q_dev, q_norms, keys, out_d, out_i, kill_flag, flat_dots,
      filt_dev;
  std::vector<uint32_t> filt_host;
  /* index-side (coarse assign + ADC scan) */
  DeviceBuf dots, sel_keys, probes, pdists, atab;
  /* cache-clustered query schedule (see kernels.h ivfpq_scan qmap) */
  DeviceBuf qcol, qmap;
  std::vector<int32_t> qcol_h, qmap_h;
  DeviceBuf rot_q, rot_norms; /* OPQ-rotated queries + their norms */
  bool in_use = false;
  ~SearchScratch() {
    if (stream) (void)hipStreamDestroy(stream);
  }
};

class IVFIndex {
 public:
  int init(int d, const IndexParams &p);
  bool trained() const { return trained_; }
  int train(const float *xt, int64_t n, hipStream_t s, std::string *err);
  int add(const float *x_host, const int64_t *vids, int64_t n,
          hipStream_t s);
  /* Lock-free-append support (realtime_mem_data.cc:57-68 publication
   * order): prepare_fast_one assigns/encodes ONE vector
   * without touching any shared structure and bails (returns 1) when
   * the target bucket would need extension (or anything else needs the
   * write-locked slow path); commit_fast_one then appends into the
   * bucket's existing capacity and publishes the new size LAST — host
   * AND the device GammaBucketDev entry — so concurrent read-locked
   * searches always see a consistent prefix. Appenders are serialized
   * by the engine's append mutex. Returns 0 ok, 1 = use slow path,
   * -1 error. */
  int prepare_fast_one(const float *vec_h, hipStream_t s,
                       int32_t *out_bucket, uint8_t *code_out,
                       float *sval_out);
  int commit_fast_one(int32_t b, int64_t vid, const float *vec_h,
                      const uint8_t *code, float sval, hipStream_t s);
  int del(int64_t vid, hipStream_t s); /* set bit 63 in the bucket slot */
  /* search: writes keys (nq x k2) into out_keys (device) */
  /* S = probe-split (out_keys must hold nq*S*k2 keys); see
   * probe_split() for the small-batch policy */
  int search(const float *q_dev, int nq, int k2, int nprobe,
             const uint32_t *bitmap_dev, bool metric_ip, hipStream_t s,
             uint64_t *out_keys_dev, const float *q_norms_dev,
             double *t_assign_ms, double *t_scan_ms, SearchScratch &sc,
             int S = 1, const int *kill_flag_dev = nullptr);
  int probe_split(int nq, int k2, int nprobe) const;
  int coarse_assign(const float *q_dev, int nq, int nprobe, bool ip,
                    const float *q_norms_dev, hipStream_t s,
                    int64_t *probes_dev, float *probe_dists_dev,
                    SearchScratch &sc);
  int64_t ntotal() const { return ntotal_; }
  const IndexParams &params() const { return params_; }
  int d() const { return d_; }
  int M() const { return M_; }
  int code_size() const { return code_size_; }
  int copy_model_to_host(float *centroids, float *codebooks,
                         hipStream_t s) const;
  bool has_opq() const { return params_.has_opq && params_.kind == IndexKind::IVFPQ; }
  const std::vector<float> &opq_R_host() const { return opq_R_host_; }
  /* y = R x for n row vectors, device to device (chunked MFMA GEMM) */
  int rotate_dev(const float *x_dev, int64_t n, float *y_dev,
                 hipStream_t s) const;
  int64_t list_size(int64_t ln) const;
  int copy_list_to_host(int64_t ln, int64_t *ids, uint8_t *codes,
                        hipStream_t s) const;
  int dump(FILE *f, hipStream_t s) const;
  int load(FILE *f, hipStream_t s);

 private:
  int kmeans_gpu(const float *x_host, int64_t n, int ncl, int niter,
                 bool spherical, std::vector<float> &cent_out,
                 hipStream_t s);
  int pq_subspace_kmeans_(const float *sub_host, int64_t n,
                          std::vector<float> &cb, hipStream_t s,
                          int seed_off);
  int update_dev_buckets(hipStream_t s);
  /* OPQ-NP training: alternate PQ fit and orthogonal Procrustes
   * (SVD), the algorithm behind faiss OPQMatrix::train (the reference
   * trains/applies it at ivfpq.cc:362-364, 470-471, 585-588) */
  int train_opq_(const float *xt, int64_t n, hipStream_t s,
                 std::string *err);
  IndexParams params_;
  bool trained_ = false;
  int d_ = 0, M_ = 0, ksub_ = 256, dsub_ = 0, code_size_ = 0, nlist_ = 0;
  int64_t ntotal_ = 0;
  DeviceBuf centroids_, cent_norms_, codebooks_;
  DeviceBuf btable_; /* pct1 B table: nlist x M x ksub f32 */
  DeviceBuf opq_R_;  /* d x d rotation, row-major: y_j = R[j] . x */
  std::vector<float> opq_R_host_;
  struct Bucket {
    std::unique_ptr<DeviceBuf> ids, data;
    /* IVFPQ: per-entry S term (see kernels.h pq_sterm); recomputable
     * from codes + the B table, so Dump stays format-compatible */
    std::unique_ptr<DeviceBuf> svals;
    long long size = 0, cap = 0;
  };
  std::vector<Bucket> buckets_;
  DeviceBuf dev_buckets_; /* GammaBucketDev[nlist] */
  bool dev_buckets_dirty_ = true;
  /* vid -> (bucket<<40 | pos); -1 = absent. Dense (vids are 0..N). */
  std::vector<int64_t> vid_loc_;
  /* scratch for add/train (grow-only; search scratch lives in
   * SearchScratch so searches can run concurrently) */
  DeviceBuf scratch_i32_, scratch_f32_;
  /* bulk-ingest staging (gk::bucket_scatter) */
  DeviceBuf scat_segs_, scat_ids_, scat_data_, scat_svals_;
  /* lock-free-append scratch (prepare_fast_one; appenders serialize) */
  DeviceBuf fast_xd_, fast_xrot_, fast_xnorm_, fast_dots_, fast_asg_,
      fast_resid_, fast_codes_, fast_sterm_;
  std::mutex bk_mu_; /* guards the one lazy write under shared lock:
                        first update_dev_buckets with no prior add */
};

/* Per-field scalar index (reference: internal/engine/table +
 * field_range_index roaring bitmaps, SURVEY 2 "Scalar table + indexes"
 * and 8f-2): term posting lists + value-sorted arrays, appended
 * incrementally as docs arrive, so a filtered search touches only the
 * matching docids instead of decoding every doc's bytes per filter.
 * Derived data: rebuilt lazily after load(), never dumped. */
struct ScalarFieldIndex {
  int64_t terms_upto = 0; /* postings cover docids [0, terms_upto) */
  int64_t range_upto = 0; /* nvals/svals cover docids [0, range_upto) */
  /* raw field bytes (or \x01-split elements for STRINGARRAY) -> ids */
  std::unordered_map<std::string, std::vector<int64_t>> postings;
  std::vector<std::pair<double, int64_t>> nvals;       /* numeric asc */
  std::vector<std::pair<std::string, int64_t>> svals;  /* string asc */
};

struct FieldMeta {
  std::string name;
  int data_type = 0; /* gamma_api DataType */
};

/* scalar filters (vearchpb Term/RangeFilter). Evaluated on
 * the host columns into an exclusion bitmap the scan kernels consume —
 * scalar INDEX acceleration (roaring/inverted) is a later row; this is
 * a linear predicate pass per request. */
struct TermFilterSpec {
  std::string field, value; /* value: terms separated by \x01 */
  int is_union = 0;
};
struct RangeFilterSpec {
  std::string field, lower, upper; /* raw binary of the field type */
  bool inc_l = false, inc_u = false;
  /* FilterOperator per filter (scalar_index_types.h:44 And=0,Or=1,Not=2;
   * engine.cc:475 casts is_union straight to it). Not + lower==upper
   * inclusive = NotEqual (bitmap_index.cc:196). */
  int is_union = 0;
};

/* The engine: one per Init() (one Vearch partition). Single vector field
 * round 1 (vector_manager multi-field merging is a later row). */
/* one additional vector field of a multi-vector table (the primary
 * field stays in the Engine's legacy members; extra fields each carry
 * their own raw store + index, sharing the table's index params —
 * vector_manager.cc:898 keeps one IndexModel per field) */
struct ExtraVecField {
  std::string name;
  int dim = 0;
  RawStore raw;
  std::unique_ptr<IVFIndex> index;
};

/* one per-field query of a multi-vector search (vector_manager.cc:851
 * dispatch); value points at nq*dim floats. min/max mirror the
 * per-field SearchCondition::IsSimilarScoreValid range
 * (gamma_common_data.h:94-96), applied to the field's candidates
 * before the docid-intersection merge. */
struct MultiVecQuery {
  std::string name;
  const float *vecs = nullptr;
  bool has_min = false, has_max = false;
  double min_score = 0, max_score = 0;
};

class Engine {
 public:
  int init(const std::string &config_json, std::string *err);
  int create_table(const std::string &name,
                   const std::vector<FieldMeta> &scalar_fields,
                   const std::string &vec_name, int dimension,
                   const std::string &index_type,
                   const std::string &index_params_json,
                   int training_threshold, std::string *err,
                   const std::vector<std::pair<std::string, int>>
                       &extra_vec_fields = {});

  /* add one doc (p_key + scalar field bytes + vector). extra_vecs:
   * one entry per extra vector field (name -> nq=1 row), required for
   * every extra field of a multi-vector table. */
  int add_doc(const std::string &p_key,
              const std::vector<std::pair<std::string, std::string>> &fields,
              const float *vec, int vec_len,
              const std::vector<MultiVecQuery> *extra_vecs = nullptr);
  int bulk_add(int64_t n, const float *vecs);
  int delete_doc(const std::string &p_key);
  int build_index(std::string *err);
  /* RebuildIndex (gamma_api.h:103): drop the trained model + lists and
   * retrain from the current raw vectors. */
  int rebuild_index(bool drop_before_rebuild, std::string *err);

  /* the hot path: batched vector search.
   * metric: 0 default, 1 L2, 2 IP. Returns 0 ok, -2 killed, <0 error.
   * xq == nullptr -> use the cached device-resident queries (bench path,
   * nq must equal the cached count). */
  int search(int nq, const float *xq, int k, int nprobe, int recall_num,
             int metric, bool brute_force, const std::string &request_id,
             int partition_id, float *out_dists, int64_t *out_ids,
             bool l2_sqrt = false,
             const std::vector<TermFilterSpec> *term_filters = nullptr,
             const std::vector<RangeFilterSpec> *range_filters = nullptr,
             std::string *filter_err = nullptr, int filter_op = 0,
             bool prelocked = false);

  /* Shared (read) lock on the engine. The C ABI takes this across
   * Search/Query/GetDoc* response assembly so the doc/table state read
   * while serializing (pkey_of, field_value, raw().host_row, bitmap)
   * cannot be reallocated by a concurrent AddOrUpdateDoc/BuildIndex
   * (which take the unique lock); pass prelocked=true to search() while
   * holding it. */
  std::shared_lock<std::shared_mutex> read_lock() const {
    return std::shared_lock<std::shared_mutex>(rw_);
  }
  /* Multi-vector search (vector_manager.cc:851-1090): one query batch
   * across several vector fields. Per field: the field's own index
   * searched to depth topn; merge: docid intersection across ALL
   * fields (seek loop :1025-1070), score = sum_j weight_j * dist_j
   * (WeightedRanker, default 1/vec_num). Output per query: up to topn
   * (docid, score) pairs — docid-ascending, or score-ordered when
   * multi_vector_rank is set (:1073-1086). Unfilled slots id = -1. */
  int search_multi(int nq, const std::vector<MultiVecQuery> &queries,
                   int topn, int nprobe, int recall_num, int metric,
                   bool brute_force, const std::string &request_id,
                   int partition_id, const std::vector<double> &weights,
                   bool rank_by_score, double *out_scores,
                   int64_t *out_ids, std::string *err,
                   const std::vector<TermFilterSpec> *term_filters = nullptr,
                   const std::vector<RangeFilterSpec> *range_filters = nullptr,
                   int filter_op = 0, bool prelocked = false);

  const std::vector<std::unique_ptr<ExtraVecField>> &extra_vec_fields()
      const {
    return extra_vecs_;
  }
  /* dimension of a named vector field (primary or extra); -1 unknown */
  int vec_dim_of(const std::string &name) const {
    if (name == vec_name_) return dim_;
    for (auto &e : extra_vecs_)
      if (e->name == name) return e->dim;
    return -1;
  }
  size_t num_vec_fields() const { return 1 + extra_vecs_.size(); }

  /* upload queries once; later search(nq, nullptr, ...) reuses them */
  int cache_queries(int nq, const float *xq);
  int cached_nq() const { return cached_nq_; }

  int dump(std::string *err);
  int load(std::string *err);

  /* Named field indexes (gamma_api.h:107-116, Engine::AddFieldIndex /
   * RemoveFieldIndex — engine.cc:1561,1648). Scalar names route onto
   * the existing ScalarFieldIndex machinery (built eagerly here, kept
   * current lazily by filtered search); the single vector field's name
   * registers against the vector index. Duplicate add and unknown
   * remove are OK no-ops, as in the reference. */
  int add_field_index(const std::string &name,
                      const std::vector<std::string> &fields,
                      const std::string &index_type,
                      const std::string &params, std::string *err);
  int remove_field_index(const std::string &name, std::string *err);
  /* Backup (gamma_api.h:104, engine.cc:1515): command 0 = create (a
   * full dump into <path>/backup); other commands are accepted no-ops
   * as in the reference's BackupThread. */
  int backup(int command, std::string *err);

  /* doc browse by scalar predicate (Engine::Query filter path,
   * engine.cc:420+): docids passing all filters, offset/limit applied */
  int filter_docids(const std::vector<TermFilterSpec> &terms,
                    const std::vector<RangeFilterSpec> &ranges, int offset,
                    int limit, std::vector<int64_t> *out, std::string *err,
                    int filter_op = 0, bool prelocked = false);

  int64_t num_docs() const {
    return max_docid_.load(std::memory_order_acquire);
  }
  IVFIndex *index() { return index_.get(); }
  RawStore &raw() { return raw_; }
  Bitmap &bitmap() { return bitmap_; }
  hipStream_t stream() { return stream_; }
  const std::string &space() const { return space_name_; }
  bool metric_ip_default() const;
  const std::string &index_type() const { return index_type_; }
  /* SetConfig/GetConfig (gamma_api.h, engine.cc:2071-2114): the
   * RocksDB cache knob is N/A on the GPU engine (reported as 0); the
   * other fields round-trip. */
  int set_config(const std::string &json);
  std::string get_config() const;
  int64_t docid_of(const std::string &p_key) const;
  const std::string &pkey_of(int64_t docid) const;
  const std::string *field_value(int64_t docid, const std::string &f) const;
  const std::vector<FieldMeta> &scalar_fields() const { return fields_; }
  const std::string &vec_field_name() const { return vec_name_; }
  int dimension() const { return dim_; }
  double last_timing[6] = {0, 0, 0, 0, 0, 0};
  int training_threshold() const { return training_threshold_; }
  std::string status_json() const;

 private:
  int flat_search_keys(RawStore &raw, int dim, const float *q_dev,
                       int nq, int k2, const float *q_norms_dev, bool ip,
                       hipStream_t s, uint64_t *out_keys_dev,
                       const uint32_t *bm, SearchScratch &sc);
  /* one field of a multi-vector search: that field's index (or FLAT
   * fallback) to depth topn, canonical scores, host output */
  int search_field_(RawStore &raw, IVFIndex *idx, int dim,
                    const float *xq, int nq, int topn, int nprobe,
                    int recall_num, bool ip, bool brute_force,
                    SearchScratch &sc, const uint32_t *bm,
                    float *host_dists, int64_t *host_ids);
  /* 1 = excluded (deleted or fails a filter); dev_out == nullptr skips
   * the device upload (filter_docids browse path) */
  int build_filter_bitmap_(const std::vector<TermFilterSpec> &terms,
                           const std::vector<RangeFilterSpec> &ranges,
                           SearchScratch &sc, const uint32_t **dev_out,
                           std::string *err, int filter_op = 0);
  std::string path_, log_dir_, space_name_;
  std::string index_type_ = "IVFPQ";
  std::string vec_name_;
  int dim_ = 0;
  int training_threshold_ = 0;
  std::vector<FieldMeta> fields_;
  std::unordered_map<std::string, StableStrCol> field_vals_;
  std::unordered_map<std::string, ScalarFieldIndex> scalar_idx_;
  std::mutex scalar_mu_; /* lazy index appends under the shared lock */
  /* user-named indexes: name -> covered field names (vector or scalar) */
  std::map<std::string, std::vector<std::string>> named_indexes_;
  const ScalarFieldIndex *extend_scalar_index_(const std::string &fname,
                                               int dt, bool want_terms,
                                               bool want_range,
                                               int64_t upto);
  int dump_to_(const std::string &dir, std::string *err);
  /* lock-free append (caller holds the SHARED lock + append_mu_):
   * 0 = done, 1 = caller must retry under the write lock, -1 error */
  int add_doc_fast_(
      const std::string &p_key,
      const std::vector<std::pair<std::string, std::string>> &fields,
      const float *vec);
  std::unordered_map<std::string, int64_t> pkey2docid_;
  StableStrCol docid2pkey_;
  mutable std::mutex pkey_mu_; /* pkey2docid_: read-locked appenders
                                  insert while Query/GetDocByID read */
  std::mutex append_mu_;       /* serializes lock-free appenders */
  RawStore raw_;
  Bitmap bitmap_;
  std::unique_ptr<IVFIndex> index_;
  std::vector<std::unique_ptr<ExtraVecField>> extra_vecs_;
  ExtraVecField *extra_vec_(const std::string &name) {
    for (auto &e : extra_vecs_)
      if (e->name == name) return e.get();
    return nullptr;
  }
  IndexParams params_;
  /* published with release AFTER all row state (columns, pkey, raw
   * vectors, bucket entry) is visible — the lock-free append contract */
  std::atomic<int64_t> max_docid_{0};
  int slow_search_time_ = 1000;  /* ms, engine.cc SetConfig field */
  int refresh_interval_ = 1000;
  std::atomic<int64_t> indexed_count_{0};
  bool table_created_ = false;
  hipStream_t stream_ = nullptr; /* mutation stream (add/build/load) */
  int cached_nq_ = 0;
  DeviceBuf cached_q_dev_; /* cache_queries() upload, read-only in search */
  mutable std::shared_mutex rw_; /* search shared / add+build exclusive */
  /* search-context pool: concurrent Search calls each take one */
  static constexpr int kMaxConcurrentSearches = 4;
  std::vector<std::unique_ptr<SearchScratch>> pool_;
  std::mutex pool_mu_;
  std::condition_variable pool_cv_;
  std::mutex timing_mu_;
  SearchScratch *acquire_scratch_();
  void release_scratch_(SearchScratch *sc);
  struct ScratchGuard {
    Engine *e;
    SearchScratch *sc;
    ~ScratchGuard() {
      if (sc) e->release_scratch_(sc);
    }
  };
};

}  // namespace vgamma
