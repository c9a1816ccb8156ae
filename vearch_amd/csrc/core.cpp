/*
 * core.cpp — engine implementation. See core.hpp for the layer map and
 * reference anchors.
 */
#include "core.hpp"

#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <algorithm>
#include <random>

#include "json.hpp"

namespace vgamma {

/* ------------------------------------------------------------ KillRegistry */
KillRegistry &KillRegistry::inst() {
  static KillRegistry k;
  return k;
}
void KillRegistry::set(const std::string &rid, int pid) {
  std::lock_guard<std::mutex> g(mu_);
  map_[{rid, pid}] = 1;
  for (auto &kv : armed_) {
    if (kv.second.first == rid && kv.second.second == pid) {
      /* a dedicated non-blocking stream: the legacy null stream would
       * wait for the very scan kernel we are trying to interrupt */
      static hipStream_t kstream = nullptr;
      if (!kstream)
        (void)hipStreamCreateWithFlags(&kstream, hipStreamNonBlocking);
      static int one = 1;
      (void)hipMemcpyAsync(kv.first, &one, 4, hipMemcpyHostToDevice,
                           kstream);
      (void)hipStreamSynchronize(kstream);
    }
  }
}

void KillRegistry::arm(const std::string &rid, int pid, int *dev_flag) {
  std::lock_guard<std::mutex> g(mu_);
  armed_[dev_flag] = {rid, pid};
}

void KillRegistry::disarm(int *dev_flag) {
  std::lock_guard<std::mutex> g(mu_);
  armed_.erase(dev_flag);
}
void KillRegistry::del(const std::string &rid, int pid) {
  std::lock_guard<std::mutex> g(mu_);
  map_.erase({rid, pid});
}
bool KillRegistry::killed(const std::string &rid, int pid) {
  if (rid.empty()) return false;
  std::lock_guard<std::mutex> g(mu_);
  return map_.count({rid, pid}) > 0;
}

/* --------------------------------------------------------------- DeviceBuf */
int DeviceBuf::reserve(size_t bytes) {
  if (bytes <= bytes_) return 0;
  void *np = nullptr;
  if (hipMalloc(&np, bytes) != hipSuccess) return -1;
  if (p_) (void)hipFree(p_);
  p_ = np;
  bytes_ = bytes;
  return 0;
}
void DeviceBuf::free() {
  if (p_) (void)hipFree(p_);
  p_ = nullptr;
  bytes_ = 0;
}

void DeviceBuf::steal(DeviceBuf &other) {
  free();
  p_ = other.p_;
  bytes_ = other.bytes_;
  other.p_ = nullptr;
  other.bytes_ = 0;
}

/* ---------------------------------------------------------------- RawStore */
int RawStore::init(int d) {
  d_ = d;
  return 0;
}

int RawStore::ensure_capacity(int64_t n_new, hipStream_t s) {
  const int64_t seg_n = (int64_t)1 << SEG_SHIFT;
  size_t need_segs = (size_t)((n_new + seg_n - 1) >> SEG_SHIFT);
  bool grew = false;
  while (dev_segs_.size() < need_segs) {
    void *p = nullptr;
    if (hipMalloc(&p, (size_t)seg_n * d_ * 4) != hipSuccess) return -1;
    dev_segs_.push_back(p);
    host_segs_.emplace_back();
    host_segs_.back().resize((size_t)seg_n * d_);
    grew = true;
  }
  if (grew) {
    if (seg_table_.reserve(dev_segs_.size() * sizeof(float *))) return -1;
    if (hipMemcpy(seg_table_.get(), dev_segs_.data(),
                  dev_segs_.size() * sizeof(float *),
                  hipMemcpyHostToDevice) != hipSuccess)
      return -1;
  }
  if (n_new > norms_cap_) {
    int64_t cap = std::max<int64_t>(n_new * 2, seg_n);
    DeviceBuf nb;
    if (nb.reserve((size_t)cap * 4)) return -1;
    if (norms_cap_ > 0)
      (void)hipMemcpy(nb.get(), norms_.get(), (size_t)n_ * 4,
                      hipMemcpyDeviceToDevice);
    norms_.steal(nb);
    norms_cap_ = cap;
  }
  return 0;
}

int RawStore::add(const float *x, int64_t cnt, hipStream_t s) {
  if (cnt <= 0) return 0;
  const int64_t n0 = n_.load(std::memory_order_relaxed);
  if (ensure_capacity(n0 + cnt, s)) return -1;
  const int64_t seg_n = (int64_t)1 << SEG_SHIFT;
  int64_t done = 0;
  while (done < cnt) {
    int64_t vid = n0 + done;
    int64_t seg = vid >> SEG_SHIFT;
    int64_t off = vid & (seg_n - 1);
    int64_t room = seg_n - off;
    int64_t take = std::min(room, cnt - done);
    memcpy(host_segs_[seg].data() + (size_t)off * d_, x + (size_t)done * d_,
           (size_t)take * d_ * 4);
    if (hipMemcpyAsync((char *)dev_segs_[seg] + (size_t)off * d_ * 4,
                       x + (size_t)done * d_, (size_t)take * d_ * 4,
                       hipMemcpyHostToDevice, s) != hipSuccess)
      return -1;
    if (gk::row_norms(s, (const float *)dev_segs_[seg] + (size_t)off * d_,
                      take, d_, norms_.as<float>() + vid) != hipSuccess)
      return -1;
    done += take;
  }
  if (hipStreamSynchronize(s) != hipSuccess) return -1;
  /* publish AFTER the rows + norms are device-resident (the
   * retrieve_idx_pos_ order, realtime_mem_data.cc:57-68) */
  n_.store(n0 + cnt, std::memory_order_release);
  return 0;
}

const float *RawStore::host_row(int64_t vid) const {
  const int64_t seg_n = (int64_t)1 << SEG_SHIFT;
  return host_segs_[vid >> SEG_SHIFT].data() + (size_t)(vid & (seg_n - 1)) * d_;
}

void RawStore::host_copy(int64_t start, int64_t cnt, float *out) const {
  for (int64_t i = 0; i < cnt;) {
    int64_t vid = start + i;
    const int64_t seg_n = (int64_t)1 << SEG_SHIFT;
    int64_t off = vid & (seg_n - 1);
    int64_t take = std::min(seg_n - off, cnt - i);
    memcpy(out + (size_t)i * d_, host_row(vid), (size_t)take * d_ * 4);
    i += take;
  }
}

const float *RawStore::dev_run(int64_t vid, int64_t *run_len) const {
  const int64_t seg_n = (int64_t)1 << SEG_SHIFT;
  int64_t seg = vid >> SEG_SHIFT, off = vid & (seg_n - 1);
  int64_t end = std::min(n_.load(std::memory_order_acquire),
                         (seg + 1) << SEG_SHIFT);
  *run_len = end - vid;
  return (const float *)dev_segs_[seg] + (size_t)off * d_;
}

int RawStore::dump(FILE *f) const {
  fwrite(&d_, 4, 1, f);
  int64_t n = n_.load(std::memory_order_acquire);
  fwrite(&n, 8, 1, f);
  const int64_t seg_n = (int64_t)1 << SEG_SHIFT;
  for (int64_t s = 0; s * seg_n < n; s++) {
    int64_t take = std::min(seg_n, n - s * seg_n);
    fwrite(host_segs_[s].data(), 4, (size_t)take * d_, f);
  }
  return 0;
}

int RawStore::load(FILE *f, hipStream_t s) {
  int d = 0;
  int64_t n = 0;
  if (fread(&d, 4, 1, f) != 1 || fread(&n, 8, 1, f) != 1) return -1;
  d_ = d;
  std::vector<float> buf((size_t)std::min<int64_t>(n, 65536) * d_);
  for (int64_t done = 0; done < n;) {
    int64_t take = std::min<int64_t>(65536, n - done);
    if (fread(buf.data(), 4, (size_t)take * d_, f) != (size_t)take * d_)
      return -1;
    if (add(buf.data(), take, s)) return -1;
    done += take;
  }
  return 0;
}

/* ------------------------------------------------------------------ Bitmap */
int Bitmap::ensure(int64_t nbits, hipStream_t s) {
  if (nbits <= bits_) return 0;
  int64_t words = (std::max<int64_t>(nbits, 1 << 20) + 31) / 32;
  words = std::max<int64_t>(words * 2, (int64_t)host_.size());
  size_t old_words = host_.size();
  host_.resize(words, 0);
  DeviceBuf nd;
  if (nd.reserve((size_t)words * 4)) return -1;
  (void)hipMemset(nd.get(), 0, (size_t)words * 4);
  if (old_words)
    (void)hipMemcpy(nd.get(), dev_.get(), old_words * 4,
                    hipMemcpyDeviceToDevice);
  dev_.steal(nd);
  bits_ = words * 32;
  return 0;
}
int Bitmap::set(int64_t vid, hipStream_t s) {
  if (ensure(vid + 1, s)) return -1;
  if (!((host_[vid >> 5] >> (vid & 31)) & 1u)) set_count_++;
  host_[vid >> 5] |= 1u << (vid & 31);
  /* mirror the single word to device */
  if (hipMemcpy((uint32_t *)dev_.get() + (vid >> 5), &host_[vid >> 5], 4,
                hipMemcpyHostToDevice) != hipSuccess)
    return -1;
  return 0;
}
bool Bitmap::test(int64_t vid) const {
  if (vid < 0 || vid >= bits_) return false;
  return (host_[vid >> 5] >> (vid & 31)) & 1u;
}
int64_t Bitmap::popcount() const {
  int64_t c = 0;
  for (uint32_t w : host_) c += __builtin_popcount(w);
  return c;
}
int Bitmap::dump(FILE *f) const {
  int64_t words = (int64_t)host_.size();
  fwrite(&words, 8, 1, f);
  fwrite(host_.data(), 4, host_.size(), f);
  return 0;
}
int Bitmap::load(FILE *f, hipStream_t s) {
  int64_t words = 0;
  if (fread(&words, 8, 1, f) != 1) return -1;
  std::vector<uint32_t> tmp(words);
  if (words && fread(tmp.data(), 4, words, f) != (size_t)words) return -1;
  if (ensure(words * 32, s)) return -1;
  std::copy(tmp.begin(), tmp.end(), host_.begin());
  set_count_ = 0;
  for (uint32_t w : host_) set_count_ += __builtin_popcount(w);
  if (words)
    (void)hipMemcpy(dev_.get(), host_.data(), (size_t)words * 4,
              hipMemcpyHostToDevice);
  return 0;
}

/* ------------------------------------------------------------- IndexParams */
int IndexParams::parse(const std::string &json, std::string *err) {
  if (json.empty()) return 0;
  gjson::Value v;
  if (!gjson::parse(json, v)) {
    if (err) *err = "parse index params error: " + json;
    return -1;
  }
  int x;
  if (v.get_int("ncentroids", x) && x > 0) ncentroids = x;
  if (v.get_int("nsubvector", x) && x > 0) nsubvector = x;
  if (v.get_int("nbits_per_idx", x) && x > 0) nbits = x;
  if (v.get_int("nprobe", x) && x > 0) nprobe = x;
  if (v.get_int("bucket_init_size", x) && x > 0) bucket_init_size = x;
  if (v.get_int("bucket_max_size", x) && x > 0) bucket_max_size = x;
  if (v.get_int("training_threshold", x) && x > 0) training_threshold = x;
  std::string mt;
  if (v.get_str("metric_type", mt)) {
    if (!strcasecmp(mt.c_str(), "L2")) metric_ip = false;
    else if (!strcasecmp(mt.c_str(), "InnerProduct")) metric_ip = true;
    else {
      if (err) *err = "invalid metric_type = " + mt;
      return -1;
    }
  }
  if (nbits != 8) {
    if (err) *err = "only nbits_per_idx=8 supported";
    return -1;
  }
  if (v.has("hnsw")) {
    if (err) *err = "hnsw coarse quantizer not supported (out of scope)";
    return -1;
  }
  if (v.has("opq")) {
    /* {"opq": {"nsubvector": N}} (ivfpq.h:1202-1212) */
    const gjson::Value *o = v.get("opq");
    if (!o || o->type != gjson::Value::OBJ) {
      if (err) *err = "invalid opq params";
      return -1;
    }
    int on = 0;
    if (!o->get_int("nsubvector", on) || on <= 0) {
      if (err) *err = "invalid opq_nsubvector";
      return -1;
    }
    has_opq = true;
    opq_nsubvector = on;
  }
  return 0;
}

/* ---------------------------------------------------------------- IVFIndex */
int IVFIndex::init(int d, const IndexParams &p) {
  params_ = p;
  d_ = d;
  if (p.has_opq) {
    /* mirror the reference's validation (ivfpq.cc:169-174) */
    if (p.kind != IndexKind::IVFPQ) return -1;
    if (p.opq_nsubvector <= 0 || d % p.opq_nsubvector != 0) return -1;
  }
  nlist_ = p.ncentroids;
  if (params_.kind == IndexKind::IVFPQ) {
    M_ = p.nsubvector > 0 ? p.nsubvector : d / 2; /* ivfpq.cc:122-124 */
    if (M_ <= 0 || d % M_ != 0 || M_ % 4 != 0) return -1;
    dsub_ = d / M_;
    code_size_ = M_;
  } else {
    M_ = 0;
    dsub_ = 0;
    code_size_ = d * 4;
  }
  buckets_.resize(nlist_);
  return 0;
}

int IVFIndex::kmeans_gpu(const float *x_host, int64_t n, int ncl, int niter,
                         bool spherical, std::vector<float> &cent_out,
                         hipStream_t s) {
  /* Lloyd, faiss-Clustering style (ivfpq.cc:188-191): random-sample init,
   * largest-cluster split for empties, optional spherical renorm. */
  DeviceBuf xd, xnorm, cd, cnorm, dots, asg;
  GAMMA_CHECK(xd.reserve((size_t)n * d_ * 4) ? hipErrorOutOfMemory
                                             : hipSuccess);
  GAMMA_CHECK(hipMemcpy(xd.get(), x_host, (size_t)n * d_ * 4,
                        hipMemcpyHostToDevice));
  GAMMA_CHECK(xnorm.reserve((size_t)n * 4) ? hipErrorOutOfMemory
                                           : hipSuccess);
  GAMMA_CHECK(gk::row_norms(s, xd.as<float>(), n, d_, xnorm.as<float>()));
  const int64_t chunk = 16384;
  GAMMA_CHECK(dots.reserve((size_t)std::min(n, chunk) * ncl * 4)
                  ? hipErrorOutOfMemory
                  : hipSuccess);
  GAMMA_CHECK(asg.reserve((size_t)n * 4) ? hipErrorOutOfMemory : hipSuccess);
  GAMMA_CHECK(cd.reserve((size_t)ncl * d_ * 4) ? hipErrorOutOfMemory
                                               : hipSuccess);
  GAMMA_CHECK(cnorm.reserve((size_t)ncl * 4) ? hipErrorOutOfMemory
                                             : hipSuccess);

  cent_out.resize((size_t)ncl * d_);
  std::mt19937_64 rng(42);
  std::vector<int64_t> perm(n);
  for (int64_t i = 0; i < n; i++) perm[i] = i;
  for (int64_t i = 0; i < std::min<int64_t>(ncl, n); i++) {
    std::swap(perm[i], perm[i + (int64_t)(rng() % (uint64_t)(n - i))]);
  }
  for (int c = 0; c < ncl; c++) {
    int64_t src = perm[c % n];
    memcpy(cent_out.data() + (size_t)c * d_, x_host + (size_t)src * d_,
           (size_t)d_ * 4);
  }
  auto renorm = [&](std::vector<float> &cent) {
    if (!spherical) return;
    for (int c = 0; c < ncl; c++) {
      double nn = 0;
      float *row = cent.data() + (size_t)c * d_;
      for (int j = 0; j < d_; j++) nn += (double)row[j] * row[j];
      float inv = (float)(1.0 / std::max(sqrt(nn), 1e-20));
      for (int j = 0; j < d_; j++) row[j] *= inv;
    }
  };
  renorm(cent_out);

  std::vector<int32_t> assign_h(n);
  std::vector<double> sums((size_t)ncl * d_);
  std::vector<int64_t> counts(ncl);

  for (int it = 0; it < niter; it++) {
    GAMMA_CHECK(hipMemcpy(cd.get(), cent_out.data(), (size_t)ncl * d_ * 4,
                          hipMemcpyHostToDevice));
    GAMMA_CHECK(gk::row_norms(s, cd.as<float>(), ncl, d_,
                              cnorm.as<float>()));
    for (int64_t r0 = 0; r0 < n; r0 += chunk) {
      int64_t rn = std::min(chunk, n - r0);
      GAMMA_CHECK(gk::dots_mfma(s, xd.as<float>() + (size_t)r0 * d_, (int)rn,
                                cd.as<float>(), ncl, d_, dots.as<float>()));
      GAMMA_CHECK(gk::argmin_rows(s, rn, ncl, dots.as<float>(),
                                  xnorm.as<float>() + r0, cnorm.as<float>(),
                                  !spherical, asg.as<int32_t>() + r0));
    }
    GAMMA_CHECK(hipMemcpy(assign_h.data(), asg.get(), (size_t)n * 4,
                          hipMemcpyDeviceToHost));
    std::fill(sums.begin(), sums.end(), 0.0);
    std::fill(counts.begin(), counts.end(), 0);
    for (int64_t i = 0; i < n; i++) {
      int c = assign_h[i];
      const float *row = x_host + (size_t)i * d_;
      double *srow = sums.data() + (size_t)c * d_;
      for (int j = 0; j < d_; j++) srow[j] += row[j];
      counts[c]++;
    }
    for (int c = 0; c < ncl; c++) {
      if (counts[c] == 0) { /* split the largest cluster */
        int big = (int)(std::max_element(counts.begin(), counts.end()) -
                        counts.begin());
        if (counts[big] < 2) continue;
        for (int j = 0; j < d_; j++) {
          double mean = sums[(size_t)big * d_ + j] / counts[big];
          double eps = 1e-5 * (1.0 + fabs(mean));
          sums[(size_t)c * d_ + j] = (mean + eps) * (counts[big] / 2);
          sums[(size_t)big * d_ + j] =
              (mean - eps) * (counts[big] - counts[big] / 2);
        }
        counts[c] = counts[big] / 2;
        counts[big] -= counts[c];
      }
    }
    for (int c = 0; c < ncl; c++) {
      float *row = cent_out.data() + (size_t)c * d_;
      for (int j = 0; j < d_; j++)
        row[j] = (float)(sums[(size_t)c * d_ + j] /
                         std::max<int64_t>(counts[c], 1));
    }
    renorm(cent_out);
  }
  return 0;
}

int IVFIndex::rotate_dev(const float *x_dev, int64_t n, float *y_dev,
                         hipStream_t s) const {
  /* y_i = R x_i as a row-GEMM: dots(x_i, R_j) = (R x_i)_j */
  const int64_t chunk = 16384;
  for (int64_t r0 = 0; r0 < n; r0 += chunk) {
    int64_t rn = std::min(chunk, n - r0);
    if (gk::dots_mfma(s, x_dev + (size_t)r0 * d_, (int)rn,
                      opq_R_.as<float>(), d_, d_,
                      y_dev + (size_t)r0 * d_) != hipSuccess)
      return -1;
  }
  return 0;
}

namespace {
/* One-sided Jacobi SVD of a dense dxd matrix (double, in place):
 * orthogonalizes A's columns with Givens rotations accumulated in V,
 * so A_in = U diag(sigma) V^T with U = normalized columns of A_out.
 * d <= ~1k, used once per OPQ iteration at train time. */
void jacobi_svd(std::vector<double> &A, int d, std::vector<double> &V) {
  V.assign((size_t)d * d, 0.0);
  for (int i = 0; i < d; i++) V[(size_t)i * d + i] = 1.0;
  const double eps = 1e-12;
  for (int sweep = 0; sweep < 30; sweep++) {
    double off = 0.0;
    for (int p = 0; p < d - 1; p++) {
      for (int q = p + 1; q < d; q++) {
        double app = 0, aqq = 0, apq = 0;
        for (int i = 0; i < d; i++) {
          double x = A[(size_t)i * d + p], y = A[(size_t)i * d + q];
          app += x * x;
          aqq += y * y;
          apq += x * y;
        }
        off += apq * apq;
        if (fabs(apq) < eps * sqrt(app * aqq) || apq == 0.0) continue;
        double tau = (aqq - app) / (2.0 * apq);
        double t = (tau >= 0 ? 1.0 : -1.0) /
                   (fabs(tau) + sqrt(1.0 + tau * tau));
        double c = 1.0 / sqrt(1.0 + t * t), sn = c * t;
        for (int i = 0; i < d; i++) {
          double x = A[(size_t)i * d + p], y = A[(size_t)i * d + q];
          A[(size_t)i * d + p] = c * x - sn * y;
          A[(size_t)i * d + q] = sn * x + c * y;
          double vx = V[(size_t)i * d + p], vy = V[(size_t)i * d + q];
          V[(size_t)i * d + p] = c * vx - sn * vy;
          V[(size_t)i * d + q] = sn * vx + c * vy;
        }
      }
    }
    if (off < 1e-20) break;
  }
}
}  // namespace

int IVFIndex::train_opq_(const float *xt, int64_t n, hipStream_t s,
                         std::string *err) {
  const int64_t n_o = std::min<int64_t>(n, 65536);
  const int niter_opq = 8;
  const int d = d_;
  /* R init: seeded Gaussian rows, Gram-Schmidt -> orthonormal */
  std::vector<double> R((size_t)d * d);
  {
    std::mt19937_64 rng(4321);
    std::normal_distribution<double> g(0.0, 1.0);
    for (auto &x : R) x = g(rng);
    for (int i = 0; i < d; i++) {
      double *ri = R.data() + (size_t)i * d;
      for (int j = 0; j < i; j++) {
        const double *rj = R.data() + (size_t)j * d;
        double dot = 0;
        for (int t = 0; t < d; t++) dot += ri[t] * rj[t];
        for (int t = 0; t < d; t++) ri[t] -= dot * rj[t];
      }
      double nrm = 0;
      for (int t = 0; t < d; t++) nrm += ri[t] * ri[t];
      nrm = sqrt(std::max(nrm, 1e-30));
      for (int t = 0; t < d; t++) ri[t] /= nrm;
    }
  }
  DeviceBuf xd, xr, codes_d;
  if (xd.reserve((size_t)n_o * d * 4)) return -1;
  GAMMA_CHECK(hipMemcpy(xd.get(), xt, (size_t)n_o * d * 4,
                        hipMemcpyHostToDevice));
  if (xr.reserve((size_t)n_o * d * 4)) return -1;
  if (codes_d.reserve((size_t)n_o * M_)) return -1;
  if (opq_R_.reserve((size_t)d * d * 4)) return -1;
  std::vector<float> Rf((size_t)d * d);
  std::vector<float> xr_h((size_t)n_o * d);
  std::vector<uint8_t> codes_h((size_t)n_o * M_);
  std::vector<float> books((size_t)M_ * ksub_ * dsub_);
  std::vector<float> sub((size_t)n_o * dsub_);
  std::vector<float> Xt_h, Yt_h;
  DeviceBuf Xt_d, Yt_d, B_d;

  for (int it = 0; it < niter_opq; it++) {
    for (size_t i = 0; i < Rf.size(); i++) Rf[i] = (float)R[i];
    GAMMA_CHECK(hipMemcpy(opq_R_.get(), Rf.data(), Rf.size() * 4,
                          hipMemcpyHostToDevice));
    if (rotate_dev(xd.as<float>(), n_o, xr.as<float>(), s)) return -1;
    GAMMA_CHECK(hipMemcpy(xr_h.data(), xr.get(), xr_h.size() * 4,
                          hipMemcpyDeviceToHost));
    /* fit a plain (non-residual) PQ to the rotated sample — the same
     * independent-PQ objective OPQMatrix optimizes */
    for (int m = 0; m < M_; m++) {
      for (int64_t i = 0; i < n_o; i++)
        memcpy(sub.data() + (size_t)i * dsub_,
               xr_h.data() + (size_t)i * d + (size_t)m * dsub_,
               (size_t)dsub_ * 4);
      std::vector<float> cb;
      if (pq_subspace_kmeans_(sub.data(), n_o, cb, s, m + 100 * it)) {
        if (err) *err = "opq pq train failed";
        return -1;
      }
      memcpy(books.data() + (size_t)m * ksub_ * dsub_, cb.data(),
             (size_t)ksub_ * dsub_ * 4);
    }
    if (codebooks_.reserve(books.size() * 4)) return -1;
    GAMMA_CHECK(hipMemcpy(codebooks_.get(), books.data(),
                          books.size() * 4, hipMemcpyHostToDevice));
    GAMMA_CHECK(gk::pq_encode(s, n_o, d, M_, ksub_, xr.as<float>(),
                              codebooks_.as<float>(),
                              codes_d.as<uint8_t>()));
    GAMMA_CHECK(hipMemcpy(codes_h.data(), codes_d.get(), codes_h.size(),
                          hipMemcpyDeviceToHost));
    /* Y = decode(codes); B = X^T Y via a GEMM on transposed layouts */
    Xt_h.assign((size_t)d * n_o, 0.f);
    Yt_h.assign((size_t)d * n_o, 0.f);
    for (int64_t i = 0; i < n_o; i++) {
      for (int t = 0; t < d; t++)
        Xt_h[(size_t)t * n_o + i] = xt[(size_t)i * d + t];
      for (int m = 0; m < M_; m++) {
        const float *cw =
            books.data() +
            ((size_t)m * ksub_ + codes_h[(size_t)i * M_ + m]) * dsub_;
        for (int t = 0; t < dsub_; t++)
          Yt_h[(size_t)(m * dsub_ + t) * n_o + i] = cw[t];
      }
    }
    if (Xt_d.reserve(Xt_h.size() * 4) || Yt_d.reserve(Yt_h.size() * 4) ||
        B_d.reserve((size_t)d * d * 4))
      return -1;
    GAMMA_CHECK(hipMemcpy(Xt_d.get(), Xt_h.data(), Xt_h.size() * 4,
                          hipMemcpyHostToDevice));
    GAMMA_CHECK(hipMemcpy(Yt_d.get(), Yt_h.data(), Yt_h.size() * 4,
                          hipMemcpyHostToDevice));
    GAMMA_CHECK(gk::dots_mfma(s, Xt_d.as<float>(), d, Yt_d.as<float>(), d,
                              (int)n_o, B_d.as<float>()));
    std::vector<float> Bf((size_t)d * d);
    GAMMA_CHECK(hipMemcpy(Bf.data(), B_d.get(), Bf.size() * 4,
                          hipMemcpyDeviceToHost));
    /* orthogonal Procrustes: max tr(R B), B = U S V^T -> R = V U^T */
    std::vector<double> A(Bf.begin(), Bf.end()), V;
    jacobi_svd(A, d, V); /* A now holds U * diag(sigma) in columns */
    for (int k = 0; k < d; k++) { /* normalize columns -> U */
      double nrm = 0;
      for (int i = 0; i < d; i++) {
        double x = A[(size_t)i * d + k];
        nrm += x * x;
      }
      nrm = sqrt(std::max(nrm, 1e-30));
      for (int i = 0; i < d; i++) A[(size_t)i * d + k] /= nrm;
    }
    for (int i = 0; i < d; i++)
      for (int j = 0; j < d; j++) {
        double acc = 0;
        for (int k = 0; k < d; k++)
          acc += V[(size_t)i * d + k] * A[(size_t)j * d + k];
        R[(size_t)i * d + j] = acc;
      }
  }
  for (size_t i = 0; i < Rf.size(); i++) Rf[i] = (float)R[i];
  GAMMA_CHECK(hipMemcpy(opq_R_.get(), Rf.data(), Rf.size() * 4,
                        hipMemcpyHostToDevice));
  opq_R_host_ = Rf;
  return 0;
}

int IVFIndex::train(const float *xt, int64_t n, hipStream_t s,
                    std::string *err) {
  std::vector<float> xrot_h;
  if (has_opq()) {
    /* train R, then train the IVFPQ model in rotated space
     * (ivfpq.cc:362-364: opq_->train + xt = opq_->apply(train)) */
    if (train_opq_(xt, n, s, err)) return -1;
    DeviceBuf xd, xr;
    if (xd.reserve((size_t)n * d_ * 4) || xr.reserve((size_t)n * d_ * 4))
      return -1;
    GAMMA_CHECK(hipMemcpy(xd.get(), xt, (size_t)n * d_ * 4,
                          hipMemcpyHostToDevice));
    if (rotate_dev(xd.as<float>(), n, xr.as<float>(), s)) return -1;
    xrot_h.resize((size_t)n * d_);
    GAMMA_CHECK(hipMemcpy(xrot_h.data(), xr.get(), xrot_h.size() * 4,
                          hipMemcpyDeviceToHost));
    xt = xrot_h.data();
  }
  std::vector<float> cent;
  if (kmeans_gpu(xt, n, nlist_, 10, params_.metric_ip, cent, s)) {
    if (err) *err = "coarse k-means failed";
    return -1;
  }
  if (centroids_.reserve((size_t)nlist_ * d_ * 4)) return -1;
  if (hipMemcpy(centroids_.get(), cent.data(), (size_t)nlist_ * d_ * 4,
                hipMemcpyHostToDevice) != hipSuccess)
    return -1;
  if (cent_norms_.reserve((size_t)nlist_ * 4)) return -1;
  if (gk::row_norms(s, centroids_.as<float>(), nlist_, d_,
                    cent_norms_.as<float>()) != hipSuccess)
    return -1;

  if (params_.kind == IndexKind::IVFPQ) {
    /* residuals of the training set under the final centroids */
    DeviceBuf xd, dots, asg, resid;
    if (xd.reserve((size_t)n * d_ * 4)) return -1;
    GAMMA_CHECK(hipMemcpy(xd.get(), xt, (size_t)n * d_ * 4,
                          hipMemcpyHostToDevice));
    DeviceBuf xnorm;
    xnorm.reserve((size_t)n * 4);
    GAMMA_CHECK(gk::row_norms(s, xd.as<float>(), n, d_, xnorm.as<float>()));
    const int64_t chunk = 16384;
    dots.reserve((size_t)std::min(n, chunk) * nlist_ * 4);
    asg.reserve((size_t)n * 4);
    for (int64_t r0 = 0; r0 < n; r0 += chunk) {
      int64_t rn = std::min(chunk, n - r0);
      if (gk::dots_mfma(s, xd.as<float>() + (size_t)r0 * d_, (int)rn,
                        centroids_.as<float>(), nlist_, d_,
                        dots.as<float>()) != hipSuccess)
        return -1;
      if (gk::argmin_rows(s, rn, nlist_, dots.as<float>(),
                          xnorm.as<float>() + r0, cent_norms_.as<float>(),
                          !params_.metric_ip,
                          asg.as<int32_t>() + r0) != hipSuccess)
        return -1;
    }
    resid.reserve((size_t)n * d_ * 4);
    if (gk::residuals(s, n, d_, xd.as<float>(), centroids_.as<float>(),
                      asg.as<int32_t>(), resid.as<float>()) != hipSuccess)
      return -1;
    std::vector<float> resid_h((size_t)n * d_);
    GAMMA_CHECK(hipMemcpy(resid_h.data(), resid.get(), (size_t)n * d_ * 4,
                          hipMemcpyDeviceToHost));

    /* per-subspace k-means, ksub=256, niter=25 (faiss PQ default) */
    std::vector<float> books((size_t)M_ * ksub_ * dsub_);
    std::vector<float> sub((size_t)n * dsub_);
    for (int m = 0; m < M_; m++) {
      for (int64_t i = 0; i < n; i++)
        memcpy(sub.data() + (size_t)i * dsub_,
               resid_h.data() + (size_t)i * d_ + (size_t)m * dsub_,
               (size_t)dsub_ * 4);
      std::vector<float> cb;
      if (pq_subspace_kmeans_(sub.data(), n, cb, s, m)) {
        if (err) *err = "pq train failed";
        return -1;
      }
      memcpy(books.data() + (size_t)m * ksub_ * dsub_, cb.data(),
             (size_t)ksub_ * dsub_ * 4);
    }
    if (codebooks_.reserve((size_t)M_ * ksub_ * dsub_ * 4)) return -1;
    GAMMA_CHECK(hipMemcpy(codebooks_.get(), books.data(),
                          (size_t)M_ * ksub_ * dsub_ * 4,
                          hipMemcpyHostToDevice));
    if (btable_.reserve((size_t)nlist_ * M_ * ksub_ * 4)) return -1;
    if (gk::pq_tables_b(s, d_, M_, nlist_, centroids_.as<float>(),
                        codebooks_.as<float>(),
                        btable_.as<float>()) != hipSuccess)
      return -1;
  }
  trained_ = true;
  return 0;
}

/* per-subspace k-means via the pq_encode kernel as the assign step */
int IVFIndex::pq_subspace_kmeans_(const float *sub_host, int64_t n,
                                  std::vector<float> &cb, hipStream_t s,
                                  int seed_off) {
  const int niter = 25;
  DeviceBuf xd, cbd, codes;
  if (xd.reserve((size_t)n * dsub_ * 4)) return -1;
  (void)hipMemcpy(xd.get(), sub_host, (size_t)n * dsub_ * 4,
            hipMemcpyHostToDevice);
  if (cbd.reserve((size_t)ksub_ * dsub_ * 4)) return -1;
  if (codes.reserve((size_t)n)) return -1;

  cb.resize((size_t)ksub_ * dsub_);
  std::mt19937_64 rng(42 + seed_off);
  std::vector<int64_t> perm(n);
  for (int64_t i = 0; i < n; i++) perm[i] = i;
  for (int64_t i = 0; i < std::min<int64_t>(ksub_, n); i++)
    std::swap(perm[i], perm[i + (int64_t)(rng() % (uint64_t)(n - i))]);
  for (int c = 0; c < ksub_; c++)
    memcpy(cb.data() + (size_t)c * dsub_,
           sub_host + (size_t)perm[c % n] * dsub_, (size_t)dsub_ * 4);

  std::vector<uint8_t> codes_h(n);
  std::vector<double> sums((size_t)ksub_ * dsub_);
  std::vector<int64_t> counts(ksub_);
  for (int it = 0; it < niter; it++) {
    (void)hipMemcpy(cbd.get(), cb.data(), (size_t)ksub_ * dsub_ * 4,
              hipMemcpyHostToDevice);
    if (gk::pq_encode(s, n, dsub_, 1, ksub_, xd.as<float>(),
                      cbd.as<float>(), codes.as<uint8_t>()) != hipSuccess)
      return -1;
    (void)hipMemcpy(codes_h.data(), codes.get(), (size_t)n, hipMemcpyDeviceToHost);
    std::fill(sums.begin(), sums.end(), 0.0);
    std::fill(counts.begin(), counts.end(), 0);
    for (int64_t i = 0; i < n; i++) {
      int c = codes_h[i];
      const float *row = sub_host + (size_t)i * dsub_;
      for (int j = 0; j < dsub_; j++) sums[(size_t)c * dsub_ + j] += row[j];
      counts[c]++;
    }
    for (int c = 0; c < ksub_; c++) {
      if (counts[c] == 0) {
        int big = (int)(std::max_element(counts.begin(), counts.end()) -
                        counts.begin());
        if (counts[big] < 2) continue;
        for (int j = 0; j < dsub_; j++) {
          double mean = sums[(size_t)big * dsub_ + j] / counts[big];
          double eps = 1e-5 * (1.0 + fabs(mean));
          sums[(size_t)c * dsub_ + j] = (mean + eps) * (counts[big] / 2);
          sums[(size_t)big * dsub_ + j] =
              (mean - eps) * (counts[big] - counts[big] / 2);
        }
        counts[c] = counts[big] / 2;
        counts[big] -= counts[c];
      }
    }
    for (int c = 0; c < ksub_; c++)
      for (int j = 0; j < dsub_; j++)
        cb[(size_t)c * dsub_ + j] =
            (float)(sums[(size_t)c * dsub_ + j] /
                    std::max<int64_t>(counts[c], 1));
  }
  return 0;
}

int IVFIndex::update_dev_buckets(hipStream_t s) {
  if (!dev_buckets_dirty_) return 0;
  /* every mutator (add/load) re-uploads before releasing the write
   * lock, so under concurrent read-locked searches this only fires on
   * the first search of an empty index — serialize that one case */
  std::lock_guard<std::mutex> lk(bk_mu_);
  if (!dev_buckets_dirty_) return 0;
  std::vector<GammaBucketDev> h(nlist_);
  for (int i = 0; i < nlist_; i++) {
    h[i].ids = buckets_[i].ids ? buckets_[i].ids->as<uint32_t>() : nullptr;
    h[i].data = buckets_[i].data ? buckets_[i].data->get() : nullptr;
    h[i].svals =
        buckets_[i].svals ? buckets_[i].svals->as<float>() : nullptr;
    h[i].size = buckets_[i].size;
  }
  if (dev_buckets_.reserve(nlist_ * sizeof(GammaBucketDev))) return -1;
  if (hipMemcpy(dev_buckets_.get(), h.data(),
                nlist_ * sizeof(GammaBucketDev),
                hipMemcpyHostToDevice) != hipSuccess)
    return -1;
  dev_buckets_dirty_ = false;
  return 0;
}

int IVFIndex::add(const float *x_host, const int64_t *vids, int64_t n,
                  hipStream_t s) {
  if (!trained_ || n <= 0) return trained_ ? 0 : -1;
  const int64_t chunk = 65536;
  const size_t entry =
      params_.kind == IndexKind::IVFPQ ? (size_t)code_size_ : (size_t)d_ * 4;

  DeviceBuf xd, xnorm, dots, asg, resid, codes, sterm;
  std::vector<int32_t> asg_h(std::min(n, chunk));
  std::vector<uint8_t> codes_h;
  std::vector<float> sterm_h;

  DeviceBuf xrot;
  for (int64_t c0 = 0; c0 < n; c0 += chunk) {
    int64_t cn = std::min(chunk, n - c0);
    if (xd.reserve((size_t)cn * d_ * 4)) return -1;
    (void)hipMemcpy(xd.get(), x_host + (size_t)c0 * d_, (size_t)cn * d_ * 4,
              hipMemcpyHostToDevice);
    const float *xin = xd.as<float>();
    if (has_opq()) { /* encode in rotated space (ivfpq.cc:470-471) */
      if (xrot.reserve((size_t)cn * d_ * 4)) return -1;
      if (rotate_dev(xin, cn, xrot.as<float>(), s)) return -1;
      xin = xrot.as<float>();
    }
    if (xnorm.reserve((size_t)cn * 4)) return -1;
    (void)gk::row_norms(s, xin, cn, d_, xnorm.as<float>());
    const int64_t sub = 16384;
    if (dots.reserve((size_t)std::min(cn, sub) * nlist_ * 4)) return -1;
    if (asg.reserve((size_t)cn * 4)) return -1;
    for (int64_t r0 = 0; r0 < cn; r0 += sub) {
      int64_t rn = std::min(sub, cn - r0);
      if (gk::dots_mfma(s, xin + (size_t)r0 * d_, (int)rn,
                        centroids_.as<float>(), nlist_, d_,
                        dots.as<float>()) != hipSuccess)
        return -1;
      if (gk::argmin_rows(s, rn, nlist_, dots.as<float>(),
                          xnorm.as<float>() + r0, cent_norms_.as<float>(),
                          !params_.metric_ip,
                          asg.as<int32_t>() + r0) != hipSuccess)
        return -1;
    }
    (void)hipMemcpy(asg_h.data(), asg.get(), (size_t)cn * 4,
              hipMemcpyDeviceToHost);

    const uint8_t *payload_h = nullptr;
    if (params_.kind == IndexKind::IVFPQ) {
      if (resid.reserve((size_t)cn * d_ * 4)) return -1;
      if (gk::residuals(s, cn, d_, xin, centroids_.as<float>(),
                        asg.as<int32_t>(), resid.as<float>()) != hipSuccess)
        return -1;
      if (codes.reserve((size_t)cn * code_size_)) return -1;
      if (gk::pq_encode(s, cn, d_, M_, ksub_, resid.as<float>(),
                        codebooks_.as<float>(),
                        codes.as<uint8_t>()) != hipSuccess)
        return -1;
      codes_h.resize((size_t)cn * code_size_);
      (void)hipMemcpy(codes_h.data(), codes.get(), (size_t)cn * code_size_,
                hipMemcpyDeviceToHost);
      /* per-vector S term against the vector's own list's B table */
      if (sterm.reserve((size_t)cn * 4)) return -1;
      if (gk::pq_sterm(s, cn, M_, nlist_, codes.as<uint8_t>(),
                       asg.as<int32_t>(), 0, btable_.as<float>(),
                       sterm.as<float>()) != hipSuccess)
        return -1;
      sterm_h.resize(cn);
      (void)hipMemcpy(sterm_h.data(), sterm.get(), (size_t)cn * 4,
                hipMemcpyDeviceToHost);
      payload_h = codes_h.data();
    } else {
      payload_h = (const uint8_t *)(x_host + (size_t)c0 * d_);
    }

    /* group by bucket (AddKeys analog, realtime_mem_data.cc) */
    const bool pq = params_.kind == IndexKind::IVFPQ;
    struct Group {
      std::vector<uint32_t> ids;
      std::vector<uint8_t> data;
      std::vector<float> svals;
    };
    std::map<int32_t, Group> groups;
    for (int64_t i = 0; i < cn; i++) {
      int32_t b = asg_h[i];
      if (b < 0 || b >= nlist_) b = (int32_t)(vids[c0 + i] % nlist_);
      auto &g = groups[b];
      g.ids.push_back((uint32_t)vids[c0 + i]); /* vid < 2^31 enforced */
      size_t off = g.data.size();
      g.data.resize(off + entry);
      memcpy(g.data.data() + off, payload_h + (size_t)i * entry, entry);
      if (pq) g.svals.push_back(sterm_h[i]);
    }
    /* capacity pass, then ONE staged upload + scatter kernel for the
     * whole chunk (3 tiny hipMemcpys per bucket per chunk were the
     * N=50M build-time bottleneck at large nlist) */
    std::vector<GammaScatterSeg> segs;
    std::vector<uint32_t> ids_cat;
    std::vector<uint8_t> data_cat;
    std::vector<float> svals_cat;
    segs.reserve(groups.size());
    ids_cat.reserve(cn);
    data_cat.reserve((size_t)cn * entry);
    if (pq) svals_cat.reserve(cn);
    for (auto &kv : groups) {
      Bucket &bk = buckets_[kv.first];
      int64_t add_n = (int64_t)kv.second.ids.size();
      if (bk.size + add_n > bk.cap) {
        long long ncap =
            std::max<long long>({(long long)params_.bucket_init_size,
                                 bk.cap * 2, bk.size + add_n});
        ncap = std::min<long long>(
            std::max<long long>(ncap, bk.size + add_n),
            std::max<long long>((long long)params_.bucket_max_size,
                                bk.size + add_n));
        auto nids = std::make_unique<DeviceBuf>();
        auto ndata = std::make_unique<DeviceBuf>();
        auto nsv = std::make_unique<DeviceBuf>();
        if (nids->reserve((size_t)ncap * 4)) return -1;
        if (ndata->reserve((size_t)ncap * entry)) return -1;
        if (pq && nsv->reserve((size_t)ncap * 4)) return -1;
        if (bk.size > 0) {
          (void)hipMemcpy(nids->get(), bk.ids->get(), (size_t)bk.size * 4,
                    hipMemcpyDeviceToDevice);
          (void)hipMemcpy(ndata->get(), bk.data->get(), (size_t)bk.size * entry,
                    hipMemcpyDeviceToDevice);
          if (pq)
            (void)hipMemcpy(nsv->get(), bk.svals->get(), (size_t)bk.size * 4,
                      hipMemcpyDeviceToDevice);
        }
        bk.ids = std::move(nids);
        bk.data = std::move(ndata);
        if (pq) bk.svals = std::move(nsv);
        bk.cap = ncap;
      }
      GammaScatterSeg sg;
      sg.ids_dst = bk.ids->as<uint32_t>() + bk.size;
      sg.data_dst = (uint8_t *)bk.data->get() + (size_t)bk.size * entry;
      sg.sval_dst = pq ? bk.svals->as<float>() + bk.size : nullptr;
      sg.src_start = (long long)ids_cat.size();
      sg.count = add_n;
      segs.push_back(sg);
      ids_cat.insert(ids_cat.end(), kv.second.ids.begin(),
                     kv.second.ids.end());
      data_cat.insert(data_cat.end(), kv.second.data.begin(),
                      kv.second.data.end());
      if (pq)
        svals_cat.insert(svals_cat.end(), kv.second.svals.begin(),
                         kv.second.svals.end());
      for (int64_t i = 0; i < add_n; i++) {
        int64_t vid = (int64_t)kv.second.ids[i];
        if (vid >= (int64_t)vid_loc_.size())
          vid_loc_.resize((size_t)vid + 1024, -1);
        vid_loc_[vid] = ((int64_t)kv.first << 40) | (bk.size + i);
      }
      bk.size += add_n;
      dev_buckets_dirty_ = true;
    }
    if (!segs.empty()) {
      if (scat_segs_.reserve(segs.size() * sizeof(GammaScatterSeg)))
        return -1;
      if (scat_ids_.reserve(ids_cat.size() * 4)) return -1;
      if (scat_data_.reserve(data_cat.size())) return -1;
      if (pq && scat_svals_.reserve(svals_cat.size() * 4)) return -1;
      GAMMA_CHECK(hipMemcpyAsync(scat_segs_.get(), segs.data(),
                                 segs.size() * sizeof(GammaScatterSeg),
                                 hipMemcpyHostToDevice, s));
      GAMMA_CHECK(hipMemcpyAsync(scat_ids_.get(), ids_cat.data(),
                                 ids_cat.size() * 4,
                                 hipMemcpyHostToDevice, s));
      GAMMA_CHECK(hipMemcpyAsync(scat_data_.get(), data_cat.data(),
                                 data_cat.size(), hipMemcpyHostToDevice,
                                 s));
      if (pq)
        GAMMA_CHECK(hipMemcpyAsync(scat_svals_.get(), svals_cat.data(),
                                   svals_cat.size() * 4,
                                   hipMemcpyHostToDevice, s));
      GAMMA_CHECK(gk::bucket_scatter(
          s, (int)segs.size(), scat_segs_.as<GammaScatterSeg>(),
          scat_ids_.as<uint32_t>(), scat_data_.as<uint8_t>(),
          pq ? scat_svals_.as<float>() : nullptr, (int)entry));
      GAMMA_CHECK(hipStreamSynchronize(s));
    }
  }
  ntotal_ += n;
  return update_dev_buckets(s);
}

int IVFIndex::prepare_fast_one(const float *vec_h, hipStream_t s,
                               int32_t *out_bucket, uint8_t *code_out,
                               float *sval_out) {
  if (!trained_) return 1;
  if (dev_buckets_dirty_) return 1; /* device table stale: slow path
                                       rebuilds it under the write lock */
  /* persistent grow-only scratch: a hipFree here would synchronize the
   * whole device and stall the concurrent read-locked searches this
   * path exists for; appenders are serialized so sharing is safe */
  DeviceBuf &xd = fast_xd_, &xrot = fast_xrot_, &xnorm = fast_xnorm_,
            &dots = fast_dots_, &asg = fast_asg_, &resid = fast_resid_,
            &codes = fast_codes_, &sterm = fast_sterm_;
  if (xd.reserve((size_t)d_ * 4)) return -1;
  GAMMA_CHECK(hipMemcpy(xd.get(), vec_h, (size_t)d_ * 4,
                        hipMemcpyHostToDevice));
  const float *xin = xd.as<float>();
  if (has_opq()) { /* encode in rotated space (ivfpq.cc:470-471) */
    if (xrot.reserve((size_t)d_ * 4)) return -1;
    if (rotate_dev(xin, 1, xrot.as<float>(), s)) return -1;
    xin = xrot.as<float>();
  }
  if (xnorm.reserve(4)) return -1;
  if (gk::row_norms(s, xin, 1, d_, xnorm.as<float>()) != hipSuccess)
    return -1;
  if (dots.reserve((size_t)nlist_ * 4)) return -1;
  if (gk::dots_mfma(s, xin, 1, centroids_.as<float>(), nlist_, d_,
                    dots.as<float>()) != hipSuccess)
    return -1;
  if (asg.reserve(4)) return -1;
  if (gk::argmin_rows(s, 1, nlist_, dots.as<float>(), xnorm.as<float>(),
                      cent_norms_.as<float>(), !params_.metric_ip,
                      asg.as<int32_t>()) != hipSuccess)
    return -1;
  GAMMA_CHECK(hipStreamSynchronize(s));
  int32_t b = -1;
  GAMMA_CHECK(hipMemcpy(&b, asg.get(), 4, hipMemcpyDeviceToHost));
  if (b < 0 || b >= nlist_) return 1; /* degenerate assign: slow path */
  if (buckets_[b].size + 1 > buckets_[b].cap) return 1; /* extension */
  if (params_.kind == IndexKind::IVFPQ) {
    if (resid.reserve((size_t)d_ * 4)) return -1;
    if (gk::residuals(s, 1, d_, xin, centroids_.as<float>(),
                      asg.as<int32_t>(), resid.as<float>()) != hipSuccess)
      return -1;
    if (codes.reserve((size_t)code_size_)) return -1;
    if (gk::pq_encode(s, 1, d_, M_, ksub_, resid.as<float>(),
                      codebooks_.as<float>(),
                      codes.as<uint8_t>()) != hipSuccess)
      return -1;
    if (sterm.reserve(4)) return -1;
    if (gk::pq_sterm(s, 1, M_, nlist_, codes.as<uint8_t>(),
                     asg.as<int32_t>(), 0, btable_.as<float>(),
                     sterm.as<float>()) != hipSuccess)
      return -1;
    GAMMA_CHECK(hipStreamSynchronize(s));
    GAMMA_CHECK(hipMemcpy(code_out, codes.get(), (size_t)code_size_,
                          hipMemcpyDeviceToHost));
    GAMMA_CHECK(hipMemcpy(sval_out, sterm.get(), 4,
                          hipMemcpyDeviceToHost));
  }
  *out_bucket = b;
  return 0;
}

int IVFIndex::commit_fast_one(int32_t b, int64_t vid, const float *vec_h,
                              const uint8_t *code, float sval,
                              hipStream_t s) {
  (void)s; /* synchronous small copies carry the ordering */
  Bucket &bk = buckets_[b];
  if (bk.size + 1 > bk.cap) return -1; /* prepare checked; appenders
                                          are serialized */
  const size_t entry = params_.kind == IndexKind::IVFPQ
                           ? (size_t)code_size_
                           : (size_t)d_ * 4;
  /* data first ... */
  if (params_.kind == IndexKind::IVFPQ) {
    GAMMA_CHECK(hipMemcpy((uint8_t *)bk.data->get() +
                              (size_t)bk.size * entry,
                          code, entry, hipMemcpyHostToDevice));
    GAMMA_CHECK(hipMemcpy(bk.svals->as<float>() + bk.size, &sval, 4,
                          hipMemcpyHostToDevice));
  } else {
    GAMMA_CHECK(hipMemcpy((uint8_t *)bk.data->get() +
                              (size_t)bk.size * entry,
                          vec_h, entry, hipMemcpyHostToDevice));
  }
  uint32_t id32 = (uint32_t)vid;
  GAMMA_CHECK(hipMemcpy(bk.ids->as<uint32_t>() + bk.size, &id32, 4,
                        hipMemcpyHostToDevice));
  /* ... size last: one aligned 8-byte write into this bucket's device
   * descriptor (concurrent scans read the old or the new size — the
   * retrieve_idx_pos_ publication, realtime_mem_data.cc:57-68) */
  long long nsz = bk.size + 1;
  char *size_dev = (char *)dev_buckets_.get() +
                   (size_t)b * sizeof(GammaBucketDev) +
                   offsetof(GammaBucketDev, size);
  GAMMA_CHECK(hipMemcpy(size_dev, &nsz, 8, hipMemcpyHostToDevice));
  bk.size = nsz;
  if (vid >= (int64_t)vid_loc_.size())
    vid_loc_.resize((size_t)vid + 1024, -1);
  vid_loc_[vid] = ((int64_t)b << 40) | (nsz - 1);
  ntotal_ += 1;
  return 0;
}

int IVFIndex::del(int64_t vid, hipStream_t s) {
  if (vid < 0 || vid >= (int64_t)vid_loc_.size() || vid_loc_[vid] < 0)
    return 0;
  int32_t bno = (int32_t)(vid_loc_[vid] >> 40);
  long long pos = vid_loc_[vid] & (((int64_t)1 << 40) - 1);
  Bucket &bk = buckets_[bno];
  /* device form: bit 31 = the kDelIdxMask bit-63 mark */
  uint32_t marked = (uint32_t)vid | 0x80000000u;
  if (hipMemcpy(bk.ids->as<uint32_t>() + pos, &marked, 4,
                hipMemcpyHostToDevice) != hipSuccess)
    return -1;
  return 0;
}

int IVFIndex::coarse_assign(const float *q_dev, int nq, int nprobe, bool ip,
                            const float *q_norms_dev, hipStream_t s,
                            int64_t *probes_dev, float *probe_dists_dev,
                            SearchScratch &sc) {
  const int64_t sub = 16384;
  if (sc.dots.reserve((size_t)std::min<int64_t>(nq, sub) * nlist_ * 4))
    return -1;
  /* measured: the chunked selector beats the full bitonic sort at
   * nlist>=2048 (fewer barriers); keep full sort for tiny nlist only */
  const bool full_sort = nlist_ <= 512;
  if (!full_sort && sc.sel_keys.reserve((size_t)nq * nprobe * 8))
    return -1;
  for (int64_t r0 = 0; r0 < nq; r0 += sub) {
    int64_t rn = std::min<int64_t>(sub, nq - r0);
    GAMMA_CHECK(gk::dots_mfma(s, q_dev + (size_t)r0 * d_, (int)rn,
                              centroids_.as<float>(), nlist_, d_,
                              sc.dots.as<float>()));
    if (full_sort) {
      GAMMA_CHECK(gk::select_rows_full(
          s, (int)rn, nlist_, nlist_, sc.dots.as<float>(),
          q_norms_dev + r0, cent_norms_.as<float>(), !ip, ip, nprobe,
          probe_dists_dev + (size_t)r0 * nprobe,
          probes_dev + (size_t)r0 * nprobe));
    } else {
      GAMMA_CHECK(gk::select_from_dots(
          s, (int)rn, nlist_, 0, nlist_, sc.dots.as<float>(),
          q_norms_dev + r0, cent_norms_.as<float>(), !ip, ip, nullptr,
          nprobe, sc.sel_keys.as<uint64_t>() + (size_t)r0 * nprobe,
          false));
    }
  }
  if (!full_sort)
    GAMMA_CHECK(gk::unpack_keys(s, (int64_t)nq * nprobe,
                                sc.sel_keys.as<uint64_t>(), ip,
                                probe_dists_dev, probes_dev));
  return 0;
}

int IVFIndex::probe_split(int nq, int k2, int nprobe) const {
  /* small batches underfill the 256 CUs with one WG per query: split
   * probes across S sub-workgroups (merged by sort_rows; S*k2 <= 2048
   * keeps the merge a single row sort) */
  if (params_.kind != IndexKind::IVFPQ) return 1;
  const char *force = getenv("GAMMA_SCAN_S"); /* perf experiments */
  if (force && atoi(force) > 0) {
    int S = atoi(force);
    while ((int64_t)S * k2 > 2048 && S > 1) S /= 2;
    return S;
  }
  int S = 1;
  while (nq * S < 1024 && S < nprobe && (int64_t)(2 * S) * k2 <= 2048)
    S *= 2;
  return S;
}

int IVFIndex::search(const float *q_dev, int nq, int k2, int nprobe,
                     const uint32_t *bitmap_dev, bool metric_ip,
                     hipStream_t s, uint64_t *out_keys_dev,
                     const float *q_norms_dev, double *t_assign_ms,
                     double *t_scan_ms, SearchScratch &sc, int S,
                     const int *kill_flag_dev) {
  if (!trained_) return -1;
  nprobe = std::min(nprobe, nlist_);
  if (nprobe > 1024) nprobe = 1024; /* selector cap (select.hpp) */
  if (sc.probes.reserve((size_t)nq * nprobe * 8)) return -1;
  if (sc.pdists.reserve((size_t)nq * nprobe * 4)) return -1;
  if (update_dev_buckets(s)) return -1;
  if (has_opq()) {
    /* coarse assign + ADC run in rotated space; the exact rerank leg
     * (Engine level) stays in raw space (ivfpq.cc:585-588, :735) */
    if (sc.rot_q.reserve((size_t)nq * d_ * 4)) return -1;
    if (sc.rot_norms.reserve((size_t)nq * 4)) return -1;
    if (rotate_dev(q_dev, nq, sc.rot_q.as<float>(), s)) return -1;
    GAMMA_CHECK(gk::row_norms(s, sc.rot_q.as<float>(), nq, d_,
                              sc.rot_norms.as<float>()));
    q_dev = sc.rot_q.as<float>();
    q_norms_dev = sc.rot_norms.as<float>();
  }

  struct Ev3 { /* RAII so error paths cannot leak events */
    hipEvent_t e[3];
    Ev3() { for (auto &x : e) (void)hipEventCreate(&x); }
    ~Ev3() { for (auto &x : e) (void)hipEventDestroy(x); }
  } ev;
  hipEvent_t e0 = ev.e[0], e1 = ev.e[1], e2 = ev.e[2];
  (void)hipEventRecord(e0, s);
  if (coarse_assign(q_dev, nq, nprobe, metric_ip, q_norms_dev, s,
                    sc.probes.as<int64_t>(), sc.pdists.as<float>(),
                    sc) != 0)
    return -1;
  (void)hipEventRecord(e1, s);
  if (params_.kind == IndexKind::IVFPQ) {
    const float *atab = nullptr;
    if (!metric_ip) {
      if (sc.atab.reserve((size_t)nq * M_ * ksub_ * 4)) return -1;
      GAMMA_CHECK(gk::pq_tables_a(s, nq, d_, M_, q_dev,
                                  codebooks_.as<float>(),
                                  sc.atab.as<float>()));
      atab = sc.atab.as<float>();
    }
    /* cache-clustered schedule (opt-in): sort the batch by first
     * probed list so workgroups touching the same lists run together
     * while those lists are L2/LLC-resident. MEASURED NEUTRAL on the
     * headline workload (scan 2.78 vs 2.72 ms with/without,
     * profiles/README round-2 notes): the scan sits at its
     * access-pattern ceiling, not on DRAM re-reads, so the schedule's
     * stream sync only adds cost. Kept behind GAMMA_QSORT=1 for
     * workloads with heavier list skew. */
    const int32_t *qmap_dev = nullptr;
    if (nq >= 2048 && getenv("GAMMA_QSORT") != nullptr) {
      if (sc.qcol.reserve((size_t)nq * 4)) return -1;
      if (sc.qmap.reserve((size_t)nq * 4)) return -1;
      GAMMA_CHECK(gk::extract_probe0(s, nq, nprobe,
                                     sc.probes.as<int64_t>(),
                                     sc.qcol.as<int32_t>()));
      sc.qcol_h.resize(nq);
      GAMMA_CHECK(hipMemcpyAsync(sc.qcol_h.data(), sc.qcol.get(),
                                 (size_t)nq * 4, hipMemcpyDeviceToHost,
                                 s));
      GAMMA_CHECK(hipStreamSynchronize(s));
      sc.qmap_h.resize(nq);
      {
        std::vector<int32_t> cnt((size_t)nlist_ + 2, 0);
        for (int i = 0; i < nq; i++) {
          int32_t ln = sc.qcol_h[i];
          if (ln < 0 || ln >= nlist_) ln = nlist_;
          cnt[ln + 1]++;
        }
        for (int i = 1; i <= nlist_ + 1; i++) cnt[i] += cnt[i - 1];
        for (int i = 0; i < nq; i++) {
          int32_t ln = sc.qcol_h[i];
          if (ln < 0 || ln >= nlist_) ln = nlist_;
          sc.qmap_h[cnt[ln]++] = i;
        }
      }
      GAMMA_CHECK(hipMemcpyAsync(sc.qmap.get(), sc.qmap_h.data(),
                                 (size_t)nq * 4, hipMemcpyHostToDevice,
                                 s));
      qmap_dev = sc.qmap.as<int32_t>();
    }
    GAMMA_CHECK(gk::ivfpq_scan(s, nq, S, d_, M_, nprobe, k2, q_dev,
                               centroids_.as<float>(),
                               codebooks_.as<float>(), atab,
                               sc.pdists.as<float>(),
                               dev_buckets_.as<GammaBucketDev>(), nlist_,
                               sc.probes.as<int64_t>(), bitmap_dev,
                               metric_ip, out_keys_dev, kill_flag_dev,
                               qmap_dev));
  } else {
    GAMMA_CHECK(gk::ivfflat_scan(s, nq, d_, nprobe, k2, q_dev,
                                 dev_buckets_.as<GammaBucketDev>(), nlist_,
                                 sc.probes.as<int64_t>(), bitmap_dev,
                                 metric_ip, out_keys_dev));
  }
  (void)hipEventRecord(e2, s);
  GAMMA_CHECK(hipEventSynchronize(e2));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  if (t_assign_ms) *t_assign_ms = ms;
  (void)hipEventElapsedTime(&ms, e1, e2);
  if (t_scan_ms) *t_scan_ms = ms;
  return 0;
}

int64_t IVFIndex::list_size(int64_t ln) const {
  if (ln < 0 || ln >= nlist_) return -1;
  return buckets_[ln].size;
}

int IVFIndex::copy_list_to_host(int64_t ln, int64_t *ids, uint8_t *codes,
                                hipStream_t s) const {
  if (ln < 0 || ln >= nlist_) return -1;
  const Bucket &bk = buckets_[ln];
  const size_t entry =
      params_.kind == IndexKind::IVFPQ ? (size_t)code_size_ : (size_t)d_ * 4;
  if (bk.size == 0) return 0;
  if (ids) {
    std::vector<uint32_t> tmp(bk.size);
    GAMMA_CHECK(hipMemcpy(tmp.data(), bk.ids->get(), (size_t)bk.size * 4,
                          hipMemcpyDeviceToHost));
    for (long long j = 0; j < bk.size; j++)
      ids[j] = (tmp[j] & 0x80000000u)
                   ? ((int64_t)(tmp[j] & 0x7fffffffu) |
                      (int64_t)((uint64_t)1 << 63))
                   : (int64_t)tmp[j];
  }
  if (codes)
    GAMMA_CHECK(hipMemcpy(codes, bk.data->get(), (size_t)bk.size * entry,
                          hipMemcpyDeviceToHost));
  return 0;
}

int IVFIndex::copy_model_to_host(float *centroids, float *codebooks,
                                 hipStream_t s) const {
  if (!trained_) return -1;
  if (centroids)
    GAMMA_CHECK(hipMemcpy(centroids, centroids_.get(),
                          (size_t)nlist_ * d_ * 4, hipMemcpyDeviceToHost));
  if (codebooks && params_.kind == IndexKind::IVFPQ)
    GAMMA_CHECK(hipMemcpy(codebooks, codebooks_.get(),
                          (size_t)M_ * ksub_ * dsub_ * 4,
                          hipMemcpyDeviceToHost));
  return 0;
}

int IVFIndex::dump(FILE *f, hipStream_t s) const {
  int tr = trained_ ? 1 : 0;
  fwrite(&tr, 4, 1, f);
  if (!trained_) return 0;
  std::vector<float> cent((size_t)nlist_ * d_);
  (void)hipMemcpy(cent.data(), centroids_.get(), cent.size() * 4,
            hipMemcpyDeviceToHost);
  fwrite(cent.data(), 4, cent.size(), f);
  if (params_.kind == IndexKind::IVFPQ) {
    std::vector<float> books((size_t)M_ * ksub_ * dsub_);
    (void)hipMemcpy(books.data(), codebooks_.get(), books.size() * 4,
              hipMemcpyDeviceToHost);
    fwrite(books.data(), 4, books.size(), f);
    if (has_opq()) /* write_opq analog (ivfpq.cc:1040) */
      fwrite(opq_R_host_.data(), 4, opq_R_host_.size(), f);
  }
  const size_t entry =
      params_.kind == IndexKind::IVFPQ ? (size_t)code_size_ : (size_t)d_ * 4;
  fwrite(&ntotal_, 8, 1, f);
  for (int i = 0; i < nlist_; i++) {
    long long sz = buckets_[i].size;
    fwrite(&sz, 8, 1, f);
    if (sz > 0) {
      std::vector<int64_t> ids(sz);
      std::vector<uint8_t> data((size_t)sz * entry);
      copy_list_to_host(i, ids.data(), nullptr, s);
      (void)hipMemcpy(data.data(), buckets_[i].data->get(), (size_t)sz * entry,
                hipMemcpyDeviceToHost);
      fwrite(ids.data(), 8, sz, f);
      fwrite(data.data(), 1, data.size(), f);
    }
  }
  return 0;
}

int IVFIndex::load(FILE *f, hipStream_t s) {
  int tr = 0;
  if (fread(&tr, 4, 1, f) != 1) return -1;
  if (!tr) return 0;
  std::vector<float> cent((size_t)nlist_ * d_);
  if (fread(cent.data(), 4, cent.size(), f) != cent.size()) return -1;
  if (centroids_.reserve(cent.size() * 4)) return -1;
  (void)hipMemcpy(centroids_.get(), cent.data(), cent.size() * 4,
            hipMemcpyHostToDevice);
  if (cent_norms_.reserve((size_t)nlist_ * 4)) return -1;
  (void)gk::row_norms(s, centroids_.as<float>(), nlist_, d_,
                cent_norms_.as<float>());
  if (params_.kind == IndexKind::IVFPQ) {
    std::vector<float> books((size_t)M_ * ksub_ * dsub_);
    if (fread(books.data(), 4, books.size(), f) != books.size()) return -1;
    if (codebooks_.reserve(books.size() * 4)) return -1;
    (void)hipMemcpy(codebooks_.get(), books.data(), books.size() * 4,
              hipMemcpyHostToDevice);
    if (btable_.reserve((size_t)nlist_ * M_ * ksub_ * 4)) return -1;
    if (gk::pq_tables_b(s, d_, M_, nlist_, centroids_.as<float>(),
                        codebooks_.as<float>(),
                        btable_.as<float>()) != hipSuccess)
      return -1;
    if (has_opq()) { /* read_opq analog (ivfpq.cc:1079-1080) */
      opq_R_host_.resize((size_t)d_ * d_);
      if (fread(opq_R_host_.data(), 4, opq_R_host_.size(), f) !=
          opq_R_host_.size())
        return -1;
      if (opq_R_.reserve(opq_R_host_.size() * 4)) return -1;
      (void)hipMemcpy(opq_R_.get(), opq_R_host_.data(),
                opq_R_host_.size() * 4, hipMemcpyHostToDevice);
    }
  }
  const size_t entry =
      params_.kind == IndexKind::IVFPQ ? (size_t)code_size_ : (size_t)d_ * 4;
  if (fread(&ntotal_, 8, 1, f) != 1) return -1;
  for (int i = 0; i < nlist_; i++) {
    long long sz = 0;
    if (fread(&sz, 8, 1, f) != 1) return -1;
    Bucket &bk = buckets_[i];
    bk.size = 0;
    bk.cap = 0;
    if (sz > 0) {
      std::vector<int64_t> ids(sz);
      std::vector<uint8_t> data((size_t)sz * entry);
      if (fread(ids.data(), 8, sz, f) != (size_t)sz) return -1;
      if (fread(data.data(), 1, data.size(), f) != data.size()) return -1;
      bk.ids = std::make_unique<DeviceBuf>();
      bk.data = std::make_unique<DeviceBuf>();
      if (bk.ids->reserve((size_t)sz * 4)) return -1;
      if (bk.data->reserve(data.size())) return -1;
      std::vector<uint32_t> ids32(sz);
      for (long long j = 0; j < sz; j++) {
        uint64_t v = (uint64_t)ids[j];
        ids32[j] = (v >> 63) ? ((uint32_t)(v & 0x7fffffffu) | 0x80000000u)
                             : (uint32_t)v;
      }
      (void)hipMemcpy(bk.ids->get(), ids32.data(), (size_t)sz * 4,
                hipMemcpyHostToDevice);
      (void)hipMemcpy(bk.data->get(), data.data(), data.size(),
                hipMemcpyHostToDevice);
      if (params_.kind == IndexKind::IVFPQ) {
        /* S terms are recomputable from codes + the B table, so the
         * dump format is unchanged; allocated here, filled by ONE
         * pq_sterm_buckets launch after update_dev_buckets below */
        bk.svals = std::make_unique<DeviceBuf>();
        if (bk.svals->reserve((size_t)sz * 4)) return -1;
      }
      bk.size = bk.cap = sz;
      for (long long j = 0; j < sz; j++) {
        int64_t vid = ids[j];
        if (!((uint64_t)vid >> 63)) {
          if (vid >= (int64_t)vid_loc_.size())
            vid_loc_.resize((size_t)vid + 1024, -1);
          vid_loc_[vid] = ((int64_t)i << 40) | j;
        }
      }
    }
  }
  dev_buckets_dirty_ = true;
  trained_ = true;
  if (update_dev_buckets(s)) return -1;
  if (params_.kind == IndexKind::IVFPQ) {
    if (gk::pq_sterm_buckets(s, nlist_,
                             M_, dev_buckets_.as<GammaBucketDev>(),
                             btable_.as<float>()) != hipSuccess)
      return -1;
    GAMMA_CHECK(hipStreamSynchronize(s));
  }
  return 0;
}

}  // namespace vgamma
