/* ref_stub.hpp — minimal faiss/gamma dependency stub for oracle/_ref.
 *
 * PURPOSE (test infrastructure ONLY — see oracle/ref_scan.c header):
 * oracle/_ref/libgammaref.so compiles the REFERENCE'S OWN scanner code
 * — extracted verbatim at build time from /root/reference by
 * oracle/ref_extract.sh, never committed — against this stub, so the
 * reference-authored control flow (QueryTables table builders, the
 * Gamma ADC scan loop, the FLAT scoring loop, the IVFFLAT scanner) can
 * be executed here and pin the hand-written oracle restatements in
 * oracle/ref_scan.c / oracle/gamma_oracle.py.
 *
 * What this stub restates (absent third-party code, faiss v1.14.1 —
 * pinned by /root/reference/cloud/env/install-dependencies.sh:38, not
 * vendored, cannot be built here):
 *   - fvec_* scalar kernels (faiss/utils/distances.h reference
 *     implementations). NOTE on rounding: these use the same explicit
 *     fmaf chains as oracle/ref_scan.c (oracle_l2sqr/oracle_ip), which
 *     mirror the GPU's k-ordered MFMA fma chain — the stub pins the
 *     reference's COMPOSITION of these kernels, not faiss's exact SIMD
 *     rounding (which no reference test pins either).
 *   - ProductQuantizer::compute_distance_table /
 *     compute_inner_prod_table (faiss/impl/ProductQuantizer.cc: per
 *     subquantizer m, per centroid j, one fvec op on the dsub slice).
 *   - CMin/CMax heaps with id tie-breaking (faiss/utils/Heap.h,
 *     faiss/utils/ordered_key_value.h: cmp/cmp2, heap_push, heap_pop,
 *     heap_replace_top, heap_reorder).
 *   - Index::compute_residual / IndexFlat::reconstruct (row copy and
 *     x - centroid, faiss/Index.cpp).
 *   - IndexIVFPQ::precompute_table() term-2 tables
 *     (faiss/IndexIVFPQ.cc): tab[key][m][j] = ||cw_mj||^2 +
 *     2 * (centroid_key_m . cw_mj), consumed by the extracted
 *     precompute_list_tables_L2 (gamma_index_ivfpq.h:256-263) through
 *     one fvec_madd with factor -2.
 * Gamma-side one-line constants/shims restated with citations:
 *   - realtime::kDelIdxMask / kRecoverIdxMask
 *     (index/realtime/realtime_mem_data.h:26-27)
 *   - RequestContext::is_killed* (c_api/api_data/request_context.h:51-99)
 *     — never killed in the harness
 *   - RetrievalContext IsValid/IsSimilarScoreValid
 *     (index/index_model.h:87-156) — delete-bitmap test, score always
 *     valid
 *   - RawVector::Gets/ScopeVectors (vector/raw_vector.h) — pointers
 *     into one contiguous base array
 *   - HeapForIP/HeapForL2 (gamma_index_flat.cc:33-34)
 */
#pragma once
#include <assert.h>
#include <math.h>
#include <stdint.h>
#include <string.h>

#include <limits>
#include <string>
#include <vector>

/* ------------------------------------------------------------- logging */
struct NullLog {
  template <typename T> NullLog &operator<<(const T &) { return *this; }
};
#define LOG(level) NullLog()

/* ------------------------------------------------------------ faiss:: */
namespace faiss {

using idx_t = int64_t;

enum MetricType { METRIC_INNER_PRODUCT = 0, METRIC_L2 = 1 };
inline bool is_similarity_metric(MetricType m) {
  return m == METRIC_INNER_PRODUCT;
}

inline uint64_t get_cycles() { return 0; }

#define FAISS_THROW_MSG(msg) \
  do { throw std::string(msg); } while (0)
#define FAISS_THROW_IF_NOT(cond) \
  do { if (!(cond)) throw std::string("FAISS_THROW_IF_NOT " #cond); } while (0)

/* ---- scalar fvec kernels (faiss/utils/distances.h reference impls;
 * fmaf chains matching oracle/ref_scan.c:53-66) ---- */
inline float fvec_L2sqr(const float *x, const float *y, size_t d) {
  float acc = 0.0f;
  for (size_t i = 0; i < d; i++) {
    float diff = x[i] - y[i];
    acc = fmaf(diff, diff, acc);
  }
  return acc;
}
inline float fvec_inner_product(const float *x, const float *y, size_t d) {
  float acc = 0.0f;
  for (size_t i = 0; i < d; i++) acc = fmaf(x[i], y[i], acc);
  return acc;
}
inline float fvec_norm_L2sqr(const float *x, size_t d) {
  float acc = 0.0f;
  for (size_t i = 0; i < d; i++) acc = fmaf(x[i], x[i], acc);
  return acc;
}
/* c[i] = a[i] + bf * b[i] (faiss fvec_madd reference impl, fma form) */
inline void fvec_madd(size_t n, const float *a, float bf, const float *b,
                      float *c) {
  for (size_t i = 0; i < n; i++) c[i] = fmaf(bf, b[i], a[i]);
}
inline int fvec_madd_and_argmin(size_t n, const float *a, float bf,
                                const float *b, float *c) {
  float vmin = std::numeric_limits<float>::infinity();
  int imin = -1;
  for (size_t i = 0; i < n; i++) {
    c[i] = fmaf(bf, b[i], a[i]);
    if (c[i] < vmin) { vmin = c[i]; imin = (int)i; }
  }
  return imin;
}
/* the *_dispatch names the reference calls resolve to the scalar refs */
inline float fvec_inner_product_dispatch(const float *x, const float *y,
                                         size_t d) {
  return fvec_inner_product(x, y, d);
}
inline float fvec_L2sqr_dispatch(const float *x, const float *y, size_t d) {
  return fvec_L2sqr(x, y, d);
}
inline void fvec_madd_dispatch(size_t n, const float *a, float bf,
                               const float *b, float *c) {
  fvec_madd(n, a, bf, b, c);
}
inline int fvec_madd_and_argmin_dispatch(size_t n, const float *a, float bf,
                                         const float *b, float *c) {
  return fvec_madd_and_argmin(n, a, bf, b, c);
}

/* ---- ordered key/value comparators (faiss/utils/ordered_key_value.h):
 * cmp strict, cmp2 breaks distance ties by id so heap order is a total
 * order ---- */
template <typename T_, typename TI_>
struct CMax {
  using T = T_;
  using TI = TI_;
  static bool cmp(T a, T b) { return a > b; }
  static bool cmp2(T a1, T a2, TI b1, TI b2) {
    return (a1 > a2) || ((a1 == a2) && (b1 > b2));
  }
  static T neutral() { return std::numeric_limits<T>::max(); }
};
template <typename T_, typename TI_>
struct CMin {
  using T = T_;
  using TI = TI_;
  static bool cmp(T a, T b) { return a < b; }
  static bool cmp2(T a1, T a2, TI b1, TI b2) {
    return (a1 < a2) || ((a1 == a2) && (b1 < b2));
  }
  static T neutral() { return std::numeric_limits<T>::lowest(); }
};

/* ---- binary heaps, 1-based sift (faiss/utils/Heap.h) ---- */
template <class C>
inline void heap_pop(size_t k, typename C::T *bh_val,
                     typename C::TI *bh_ids) {
  bh_val--; /* 1-based */
  bh_ids--;
  typename C::T val = bh_val[k];
  typename C::TI id = bh_ids[k];
  size_t i = 1, i1, i2;
  while (1) {
    i1 = i << 1;
    i2 = i1 + 1;
    if (i1 > k) break;
    if ((i2 == k + 1) ||
        C::cmp2(bh_val[i1], bh_val[i2], bh_ids[i1], bh_ids[i2])) {
      if (C::cmp2(val, bh_val[i1], id, bh_ids[i1])) break;
      bh_val[i] = bh_val[i1];
      bh_ids[i] = bh_ids[i1];
      i = i1;
    } else {
      if (C::cmp2(val, bh_val[i2], id, bh_ids[i2])) break;
      bh_val[i] = bh_val[i2];
      bh_ids[i] = bh_ids[i2];
      i = i2;
    }
  }
  bh_val[i] = bh_val[k];
  bh_ids[i] = bh_ids[k];
}

template <class C>
inline void heap_push(size_t k, typename C::T *bh_val, typename C::TI *bh_ids,
                      typename C::T val, typename C::TI id) {
  bh_val--;
  bh_ids--;
  size_t i = k, i_father;
  while (i > 1) {
    i_father = i >> 1;
    if (!C::cmp2(val, bh_val[i_father], id, bh_ids[i_father])) break;
    bh_val[i] = bh_val[i_father];
    bh_ids[i] = bh_ids[i_father];
    i = i_father;
  }
  bh_val[i] = val;
  bh_ids[i] = id;
}

template <class C>
inline void heap_replace_top(size_t k, typename C::T *bh_val,
                             typename C::TI *bh_ids, typename C::T val,
                             typename C::TI id) {
  bh_val--;
  bh_ids--;
  size_t i = 1, i1, i2;
  while (1) {
    i1 = i << 1;
    i2 = i1 + 1;
    if (i1 > k) break;
    if ((i2 == k + 1) ||
        C::cmp2(bh_val[i1], bh_val[i2], bh_ids[i1], bh_ids[i2])) {
      if (C::cmp2(val, bh_val[i1], id, bh_ids[i1])) break;
      bh_val[i] = bh_val[i1];
      bh_ids[i] = bh_ids[i1];
      i = i1;
    } else {
      if (C::cmp2(val, bh_val[i2], id, bh_ids[i2])) break;
      bh_val[i] = bh_val[i2];
      bh_ids[i] = bh_ids[i2];
      i = i2;
    }
  }
  bh_val[i] = val;
  bh_ids[i] = id;
}

template <class C>
inline void heap_heapify(size_t k, typename C::T *bh_val,
                         typename C::TI *bh_ids) {
  for (size_t i = 0; i < k; i++) {
    bh_val[i] = C::neutral();
    bh_ids[i] = -1;
  }
}

template <class C>
inline void heap_reorder(size_t k, typename C::T *bh_val,
                         typename C::TI *bh_ids) {
  for (size_t i = 0, ii = 0; i < k; i++) {
    typename C::T val = bh_val[0];
    typename C::TI id = bh_ids[0];
    heap_pop<C>(k - i, bh_val, bh_ids);
    bh_val[k - ii - 1] = val;
    bh_ids[k - ii - 1] = id;
    ii++;
  }
}

/* list_no/offset packing for store_pairs (faiss/Index.h lo_build) */
inline idx_t lo_build(idx_t list_id, idx_t offset) {
  return list_id << 32 | offset;
}

/* ---- 8-bit PQ code decoder (faiss
 * impl/pq_code_distance/pq_code_distance-inl.h PQDecoder8: one byte per
 * sub-index; the reference instantiates nbits=8 only on this path,
 * gamma_index_ivfpq.cc:130 code_size = nsubvector) ---- */
struct PQDecoder8 {
  const uint8_t *code;
  explicit PQDecoder8(const uint8_t *code_in, int /*nbits*/)
      : code(code_in) {}
  uint64_t decode() { return *code++; }
};

/* ---- ProductQuantizer (fields + table builders the extracted code
 * calls; faiss/impl/ProductQuantizer.cc) ---- */
struct ProductQuantizer {
  size_t d = 0;      /* total dimension */
  size_t M = 0;      /* number of subquantizers */
  size_t nbits = 8;  /* bits per sub-index */
  size_t dsub = 0;   /* d / M */
  size_t ksub = 256; /* 1 << nbits */
  size_t code_size = 0;
  std::vector<float> centroids; /* M * ksub * dsub */

  void init(size_t d_, size_t M_) {
    d = d_;
    M = M_;
    dsub = d / M;
    ksub = 256;
    nbits = 8;
    code_size = M;
    centroids.resize(M * ksub * dsub);
  }
  const float *get_centroids(size_t m, size_t j) const {
    return centroids.data() + (m * ksub + j) * dsub;
  }
  /* tab[m*ksub+j] = || x_m - cw_{m,j} ||^2 */
  void compute_distance_table(const float *x, float *tab) const {
    for (size_t m = 0; m < M; m++)
      for (size_t j = 0; j < ksub; j++)
        tab[m * ksub + j] =
            fvec_L2sqr(x + m * dsub, get_centroids(m, j), dsub);
  }
  /* tab[m*ksub+j] = x_m . cw_{m,j} */
  void compute_inner_prod_table(const float *x, float *tab) const {
    for (size_t m = 0; m < M; m++)
      for (size_t j = 0; j < ksub; j++)
        tab[m * ksub + j] =
            fvec_inner_product(x + m * dsub, get_centroids(m, j), dsub);
  }
  /* nearest codeword per subspace (only reached on polysemous paths,
   * which the harness never enables) */
  void compute_code(const float *x, uint8_t *code) const {
    for (size_t m = 0; m < M; m++) {
      float best = std::numeric_limits<float>::infinity();
      int bi = 0;
      for (size_t j = 0; j < ksub; j++) {
        float dis = fvec_L2sqr(x + m * dsub, get_centroids(m, j), dsub);
        if (dis < best) { best = dis; bi = (int)j; }
      }
      code[m] = (uint8_t)bi;
    }
  }
};

/* ---- quantizer (IndexFlat in the reference config,
 * gamma_index_ivfpq.cc:155-166): centroid table with
 * reconstruct/compute_residual (faiss/Index.cpp) ---- */
struct Index {
  int d = 0;
  virtual ~Index() = default;
  virtual void reconstruct(idx_t, float *) const {}
};
struct IndexFlat : Index {
  std::vector<float> xb; /* nlist * d centroids */
  void reconstruct(idx_t key, float *out) const override {
    memcpy(out, xb.data() + (size_t)key * d, sizeof(float) * d);
  }
  void compute_residual(const float *x, float *residual, idx_t key) const {
    const float *c = xb.data() + (size_t)key * d;
    for (int i = 0; i < d; i++) residual[i] = x[i] - c[i];
  }
};
/* only a dynamic_cast target on the use_precomputed_table==2 path,
 * which the harness never takes (reference default is 0, documented
 * alternative 1 — ivfpq.cc:196, ivfpq.h:254) */
struct MultiIndexQuantizer : Index {
  ProductQuantizer pq;
};

struct IVFSearchParameters {
  virtual ~IVFSearchParameters() = default;
};
struct IVFPQSearchParameters : IVFSearchParameters {
  int polysemous_ht = 0;
};

/* ---- IndexIVFPQ container (the fields QueryTables copies,
 * gamma_index_ivfpq.h:120-131) ---- */
struct IndexIVFPQ {
  int d = 0;
  ProductQuantizer pq;
  bool by_residual = true;
  int use_precomputed_table = 0;
  int polysemous_ht = 0;
  IndexFlat *quantizer = nullptr;
  std::vector<float> precomputed_table;

  /* faiss IndexIVFPQ::precompute_table(): term 2 of
   * ||x - (C_key + cw)||^2 = ||x-C_key||^2 (coarse_dis)
   *   + ||cw||^2 + 2*(C_key . cw)   <- this table
   *   - 2*(x . cw)                  <- sim_table_2 at query time
   * composed exactly as faiss does: inner-prod table of the centroid,
   * then one fvec_madd with factor 2 against the codeword norms. */
  void precompute_table() {
    size_t nlist = quantizer->xb.size() / d;
    size_t MK = pq.M * pq.ksub;
    std::vector<float> r_norms(MK);
    for (size_t m = 0; m < pq.M; m++)
      for (size_t j = 0; j < pq.ksub; j++)
        r_norms[m * pq.ksub + j] =
            fvec_norm_L2sqr(pq.get_centroids(m, j), pq.dsub);
    precomputed_table.resize(nlist * MK);
    std::vector<float> centroid(d);
    for (size_t key = 0; key < nlist; key++) {
      quantizer->reconstruct((idx_t)key, centroid.data());
      float *tab = precomputed_table.data() + key * MK;
      pq.compute_inner_prod_table(centroid.data(), tab);
      fvec_madd(MK, r_norms.data(), 2.0f, tab, tab);
    }
  }
};

/* ---- result plumbing the extracted WrappedSearchResult drives
 * (faiss/impl/ResultHandler.h HeapBlockResultHandler single-query
 * semantics: threshold = heap top, add_result replaces top) ---- */
struct ResultHandler {
  float threshold = 0;
  virtual ~ResultHandler() = default;
  virtual bool add_result(float dis, idx_t idx) = 0;
};
template <class C>
struct HeapResultHandler : ResultHandler {
  size_t k;
  float *heap_dis;
  idx_t *heap_ids;
  HeapResultHandler(size_t k_, float *dis, idx_t *ids)
      : k(k_), heap_dis(dis), heap_ids(ids) {
    heap_heapify<C>(k, heap_dis, heap_ids);
    threshold = heap_dis[0];
  }
  bool add_result(float dis, idx_t idx) override {
    if (C::cmp(heap_dis[0], dis)) {
      heap_replace_top<C>(k, heap_dis, heap_ids, dis, idx);
      threshold = heap_dis[0];
      return true;
    }
    return false;
  }
  void end() { heap_reorder<C>(k, heap_dis, heap_ids); }
};

struct IDSelector {
  virtual ~IDSelector() = default;
  virtual bool is_member(idx_t id) const = 0;
};

/* base of the extracted GammaIVFFlatScanner (faiss
 * invlists/InvertedLists.h InvertedListScanner, reduced to the members
 * and virtuals the extracted code declares/overrides) */
struct InvertedListScanner {
  bool store_pairs = false;
  const IDSelector *sel = nullptr;
  bool keep_max = false;
  idx_t list_no = -1;
  size_t code_size = 0;
  InvertedListScanner(bool store_pairs_in = false,
                      const IDSelector *sel_in = nullptr)
      : store_pairs(store_pairs_in), sel(sel_in) {}
  virtual ~InvertedListScanner() = default;
  virtual void set_query(const float *query) = 0;
  virtual void set_list(idx_t list_no, float coarse_dis) = 0;
  virtual float distance_to_code(const uint8_t *code) const = 0;
  virtual size_t scan_codes(size_t list_size, const uint8_t *codes,
                            const idx_t *ids, float *simi, idx_t *idxi,
                            size_t k) const {
    (void)list_size; (void)codes; (void)ids; (void)simi; (void)idxi;
    (void)k;
    return 0;
  }
};

}  // namespace faiss

/* -------------------------------------------------------- gamma shims */
namespace realtime {
/* realtime_mem_data.h:26-27 */
constexpr int64_t kDelIdxMask = (int64_t)1 << 63;
constexpr int64_t kRecoverIdxMask = ~kDelIdxMask;
}  // namespace realtime

/* request_context.h:51-99 — the harness never kills a request */
struct RequestContext {
  template <int N>
  static bool is_killed_every(size_t) { return false; }
  static bool is_killed(const std::string &, int) { return false; }
  static bool is_killed() { return false; }
};

/* index_model.h:87-156 — delete-bitmap IsValid, score always valid */
struct RetrievalContext {
  const uint8_t *del_bitmap = nullptr; /* 1 bit per docid, 1 = deleted */
  bool IsValid(int64_t id) const {
    return !(del_bitmap && (del_bitmap[id >> 3] >> (id & 7)) & 1);
  }
  bool IsSimilarScoreValid(float) const { return true; }
};

/* vector/raw_vector.h ScopeVectors/Gets — one contiguous base array */
struct ScopeVectors {
  std::vector<const uint8_t *> ptrs;
  const uint8_t *Get(size_t i) const { return ptrs[i]; }
};
struct RawVector {
  const float *base = nullptr;
  int d = 0;
  int Gets(const std::vector<int64_t> &vids, ScopeVectors &out) const {
    for (int64_t vid : vids)
      out.ptrs.push_back(
          (const uint8_t *)(base + (size_t)vid * d));
    return 0;
  }
};

using idx_t = faiss::idx_t;
/* gamma_index_flat.cc:33-34 */
using HeapForIP = faiss::CMin<float, idx_t>;
using HeapForL2 = faiss::CMax<float, idx_t>;
