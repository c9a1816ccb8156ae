/*
 * ref_scan.c — CPU restatement (plain C + OpenMP) of the Gamma hot-path
 * scan arithmetic. TEST INFRASTRUCTURE + CPU BASELINE ONLY: this code is
 * imported by tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
 * leg. The product path (libgamma.so, HIP) never calls it and fails loudly
 * when the HIP extension is missing.
 *
 * Every floating-point accumulation here is written with an explicit
 * sequential loop and explicit fmaf() so the GPU kernels (which use the
 * same order and fmaf) are bit-identical on the same inputs.
 *
 * Reference semantics restated (file:line in /root/reference):
 *  - FLAT scan: internal/engine/index/impl/gamma_index_flat.cc:63-128
 *    (ComputeScoreBatch / FlatScanRange: per-vector fvec_L2sqr /
 *    fvec_inner_product + top-k heap; delete-bitmap filter via IsValid).
 *  - ADC table build (L2, by_residual, use_precomputed_table=0):
 *    internal/engine/index/impl/gamma_index_ivfpq.h:243-249
 *    (residual r = q - c_list; T[m][j] = ||r_m - codeword_{m,j}||^2).
 *  - ADC table build (IP): gamma_index_ivfpq.h:164-167 + 223-236
 *    (query-level T[m][j] = q_m . codeword_{m,j}; dis0 = q . c_list).
 *  - ADC list scan: gamma_index_ivfpq.h:923-953 (skip if
 *    ids[j] & kDelIdxMask (bit 63, realtime_mem_data.h:26) or delete
 *    bitmap; dis = dis0 + sum_m T[m][code_m] in m order).
 *  - IVFFLAT list scan: gamma_index_ivfflat.h:36-91 (exact fp32 distance
 *    over the probed lists' raw vectors).
 *  - Ties: this rebuild defines the total order (dist, then id) — SURVEY
 *    §8c pin (i); the reference's heap order on exact ties is
 *    scan-order-dependent and unpinned.
 *
 * Parity pinning: the reference engine cannot be compiled in this
 * container (external faiss v1.14.1 / RocksDB / CRoaring, no network —
 * SURVEY §8c). FLAT exactness is pinned by the reference's own CI gate
 * (test/test_vector_index_flat.py:95-96); IVFPQ by its recall floors
 * (test/test_vector_index_ivfpq.py:106-111). Bit-exact parity at the
 * faiss boundary is unpinned — see DESIGN.md "Parity model".
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <math.h>
#ifdef _OPENMP
#include <omp.h>
#endif

#define EXPORT __attribute__((visibility("default")))

/* kDelIdxMask, realtime_mem_data.h:26 */
static const uint64_t DEL_MASK = (uint64_t)1 << 63;

/* ---- canonical distance primitives (sequential, fmaf) ---- */

EXPORT float oracle_l2sqr(const float *x, const float *y, int d) {
  float acc = 0.0f;
  for (int i = 0; i < d; i++) {
    float diff = x[i] - y[i];
    acc = fmaf(diff, diff, acc);
  }
  return acc;
}

EXPORT float oracle_ip(const float *x, const float *y, int d) {
  float acc = 0.0f;
  for (int i = 0; i < d; i++) acc = fmaf(x[i], y[i], acc);
  return acc;
}

/* ---- (dist, id) total order helpers ----
 * key = monotone u32 image of fp32 (sign-flip trick) << 32 | id.
 * Ascending key == ascending (dist, id). For IP we invert the distance
 * bits so ascending key == descending (dist), ascending id on ties. */
static inline uint32_t f32_key(float f) {
  uint32_t u;
  memcpy(&u, &f, 4);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
static inline uint64_t make_key(float dist, uint32_t id, int metric_ip) {
  uint32_t dk = f32_key(dist);
  if (metric_ip) dk = ~dk;
  return ((uint64_t)dk << 32) | (uint64_t)id;
}

/* insert into a capacity-k max-heap of u64 keys (largest key at root) */
static inline void heap_push_or_replace(uint64_t *heap, int *size, int k,
                                        uint64_t key) {
  if (*size < k) {
    int i = (*size)++;
    heap[i] = key;
    while (i > 0) {
      int p = (i - 1) >> 1;
      if (heap[p] >= heap[i]) break;
      uint64_t t = heap[p]; heap[p] = heap[i]; heap[i] = t;
      i = p;
    }
  } else if (key < heap[0]) {
    heap[0] = key;
    int i = 0;
    for (;;) {
      int l = 2 * i + 1, r = l + 1, m = i;
      if (l < k && heap[l] > heap[m]) m = l;
      if (r < k && heap[r] > heap[m]) m = r;
      if (m == i) break;
      uint64_t t = heap[m]; heap[m] = heap[i]; heap[i] = t;
      i = m;
    }
  }
}

static int cmp_u64(const void *a, const void *b) {
  uint64_t x = *(const uint64_t *)a, y = *(const uint64_t *)b;
  return x < y ? -1 : (x > y ? 1 : 0);
}

static void emit_sorted(uint64_t *heap, int size, int k, int metric_ip,
                        float *out_dists, int64_t *out_ids) {
  qsort(heap, (size_t)size, 8, cmp_u64);
  for (int i = 0; i < k; i++) {
    if (i < size) {
      uint32_t dk = (uint32_t)(heap[i] >> 32);
      if (metric_ip) dk = ~dk;
      uint32_t u = (dk & 0x80000000u) ? (dk & 0x7fffffffu) : ~dk;
      float f; memcpy(&f, &u, 4);
      out_dists[i] = f;
      out_ids[i] = (int64_t)(uint32_t)(heap[i] & 0xffffffffu);
    } else {
      out_dists[i] = -1.0f;
      out_ids[i] = -1;
    }
  }
}

static inline int bitmap_test(const uint8_t *bm, uint64_t id) {
  return bm && ((bm[id >> 3] >> (id & 7)) & 1);
}

/* ---- FLAT scan: gamma_index_flat.cc:63-128 semantics ----
 * base: n*d fp32, vid == row. del_bitmap: 1 bit per vid (1 = deleted),
 * may be NULL. metric_ip: 0 = L2, 1 = inner product. */
EXPORT void oracle_flat_search(int64_t n, int d, const float *base,
                               int nq, const float *queries, int k,
                               const uint8_t *del_bitmap, int metric_ip,
                               float *out_dists, int64_t *out_ids) {
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
  for (int qi = 0; qi < nq; qi++) {
    const float *q = queries + (size_t)qi * d;
    uint64_t *heap = (uint64_t *)malloc((size_t)k * 8);
    int size = 0;
    for (int64_t j = 0; j < n; j++) {
      if (bitmap_test(del_bitmap, (uint64_t)j)) continue;
      float dis = metric_ip ? oracle_ip(q, base + (size_t)j * d, d)
                            : oracle_l2sqr(q, base + (size_t)j * d, d);
      heap_push_or_replace(heap, &size, k,
                           make_key(dis, (uint32_t)j, metric_ip));
    }
    emit_sorted(heap, size, k, metric_ip,
                out_dists + (size_t)qi * k, out_ids + (size_t)qi * k);
    free(heap);
  }
}

/* ---- ADC table build, L2 by-residual (gamma_index_ivfpq.h:243-249) ----
 * q: d fp32; centroid: d fp32; codebooks: M*ksub*dsub fp32;
 * out_table: M*ksub fp32. */
EXPORT void oracle_adc_table_l2(int d, int M, int ksub, const float *q,
                                const float *centroid,
                                const float *codebooks, float *out_table) {
  int dsub = d / M;
  float *r = (float *)malloc((size_t)d * 4);
  for (int i = 0; i < d; i++) r[i] = q[i] - centroid[i];
  for (int m = 0; m < M; m++) {
    const float *rm = r + m * dsub;
    for (int j = 0; j < ksub; j++) {
      const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
      float acc = 0.0f;
      for (int t = 0; t < dsub; t++) {
        float diff = rm[t] - cw[t];
        acc = fmaf(diff, diff, acc);
      }
      out_table[(size_t)m * ksub + j] = acc;
    }
  }
  free(r);
}

/* ---- ADC table build, IP (gamma_index_ivfpq.h:164-167): query-level ---- */
EXPORT void oracle_adc_table_ip(int d, int M, int ksub, const float *q,
                                const float *codebooks, float *out_table) {
  int dsub = d / M;
  for (int m = 0; m < M; m++) {
    const float *qm = q + m * dsub;
    for (int j = 0; j < ksub; j++) {
      const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
      float acc = 0.0f;
      for (int t = 0; t < dsub; t++) acc = fmaf(qm[t], cw[t], acc);
      out_table[(size_t)m * ksub + j] = acc;
    }
  }
}

/* ---- ADC scan of one list (gamma_index_ivfpq.h:923-953) ----
 * Accumulates candidates into heap (caller owns across lists). */
EXPORT void oracle_adc_scan_list(int64_t ncode, int M, int ksub,
                                 const uint8_t *codes, const int64_t *ids,
                                 const float *table, float dis0,
                                 const uint8_t *del_bitmap, int metric_ip,
                                 int k, uint64_t *heap, int *heap_size) {
  for (int64_t j = 0; j < ncode; j++) {
    uint64_t id = (uint64_t)ids[j];
    if (id & DEL_MASK) continue;                 /* h:930 */
    if (bitmap_test(del_bitmap, id)) continue;   /* h:935 IsValid */
    const uint8_t *code = codes + (size_t)j * M;
    float dis = dis0;
    for (int m = 0; m < M; m++)                  /* h:945-948 */
      dis += table[(size_t)m * ksub + code[m]];
    heap_push_or_replace(heap, heap_size, k,
                         make_key(dis, (uint32_t)id, metric_ip));
  }
}

/* ---- full IVFPQ search given a trained model + assignments ----
 * (search_preassigned, gamma_index_ivfpq.cc:730-945, parallel over queries)
 * list_offsets[nlist+1] index into flat ids/codes arrays.
 * probes: nq*nprobe int64 list numbers (may be -1 = skip, cc:639).
 * coarse_dis: nq*nprobe fp32 (used for IP dis0? no: dis0 computed from
 * centroid dot product canonically — IP dis0 = q . c, h:223-236). */
EXPORT void oracle_ivfpq_search(int nq, int d, int M, int ksub, int nlist,
                                const float *queries, const float *centroids,
                                const float *codebooks,
                                const int64_t *list_offsets,
                                const int64_t *ids, const uint8_t *codes,
                                int nprobe, const int64_t *probes,
                                const uint8_t *del_bitmap, int metric_ip,
                                int k, float *out_dists, int64_t *out_ids) {
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
  for (int qi = 0; qi < nq; qi++) {
    const float *q = queries + (size_t)qi * d;
    float *table = (float *)malloc((size_t)M * ksub * 4);
    uint64_t *heap = (uint64_t *)malloc((size_t)k * 8);
    int size = 0;
    if (metric_ip) oracle_adc_table_ip(d, M, ksub, q, codebooks, table);
    for (int p = 0; p < nprobe; p++) {
      int64_t ln = probes[(size_t)qi * nprobe + p];
      if (ln < 0 || ln >= nlist) continue;
      int64_t beg = list_offsets[ln], end = list_offsets[ln + 1];
      if (end <= beg) continue;
      float dis0 = 0.0f;
      if (metric_ip) {
        dis0 = oracle_ip(q, centroids + (size_t)ln * d, d);
      } else {
        oracle_adc_table_l2(d, M, ksub, q, centroids + (size_t)ln * d,
                            codebooks, table);
      }
      oracle_adc_scan_list(end - beg, M, ksub, codes + (size_t)beg * M,
                           ids + beg, table, dis0, del_bitmap, metric_ip, k,
                           heap, &size);
    }
    emit_sorted(heap, size, k, metric_ip,
                out_dists + (size_t)qi * k, out_ids + (size_t)qi * k);
    free(table);
    free(heap);
  }
}

/* ---- decomposed ADC tables (use_precomputed_table=1 semantics,
 * gamma_index_ivfpq.h:254-262): T[m][j] = A_q[m][j] + B_list[m][j] with
 *   A[m][j] = fmaf(-2, q_m . cw, ||cw||^2)   (query-level)
 *   B[m][j] = 2 * (c_list,m . cw)            (train-time table)
 *   dis0    = fmaf(-2, q . c, ||q||^2 + ||c||^2)  (the coarse distance)
 * All dot products / norms are sequential fmaf chains — the exact
 * arithmetic of the HIP kernels (MFMA f32 is a k-ordered fmaf chain,
 * cdna_hip_programming.md §3). ---- */

EXPORT void oracle_pct1_a_table(int d, int M, int ksub, const float *q,
                                const float *codebooks, float *out_a) {
  int dsub = d / M;
  for (int m = 0; m < M; m++) {
    const float *qm = q + m * dsub;
    for (int j = 0; j < ksub; j++) {
      const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
      float cwn = 0.0f, dot = 0.0f;
      for (int t = 0; t < dsub; t++) cwn = fmaf(cw[t], cw[t], cwn);
      for (int t = 0; t < dsub; t++) dot = fmaf(qm[t], cw[t], dot);
      out_a[(size_t)m * ksub + j] = fmaf(-2.0f, dot, cwn);
    }
  }
}

EXPORT void oracle_pct1_b_table(int d, int M, int ksub,
                                const float *centroid,
                                const float *codebooks, float *out_b) {
  int dsub = d / M;
  for (int m = 0; m < M; m++) {
    const float *cm = centroid + m * dsub;
    for (int j = 0; j < ksub; j++) {
      const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
      float dot = 0.0f;
      for (int t = 0; t < dsub; t++) dot = fmaf(cm[t], cw[t], dot);
      out_b[(size_t)m * ksub + j] = 2.0f * dot;
    }
  }
}

/* GEMM-form coarse L2 distance (= the HIP engine's probe distance):
 * fmaf(-2, dot, qn + cn), sequential dots/norms. */
EXPORT float oracle_l2_gemm_form(const float *q, const float *c, int d) {
  float qn = 0.0f, cn = 0.0f, dot = 0.0f;
  for (int t = 0; t < d; t++) qn = fmaf(q[t], q[t], qn);
  for (int t = 0; t < d; t++) cn = fmaf(c[t], c[t], cn);
  for (int t = 0; t < d; t++) dot = fmaf(q[t], c[t], dot);
  return fmaf(-2.0f, dot, qn + cn);
}

/* IVFPQ L2 search with decomposed tables; probe_dists supplies dis0
 * (pass oracle_l2_gemm_form values, or the engine's probe distances for
 * bit-exact pinning). */
EXPORT void oracle_ivfpq_search_pct1(
    int nq, int d, int M, int ksub, int nlist, const float *queries,
    const float *centroids, const float *codebooks,
    const int64_t *list_offsets, const int64_t *ids, const uint8_t *codes,
    int nprobe, const int64_t *probes, const float *probe_dists,
    const uint8_t *del_bitmap, int k, float *out_dists, int64_t *out_ids) {
  float *btabs = (float *)malloc((size_t)nlist * M * ksub * 4);
  unsigned char *bdone = (unsigned char *)calloc((size_t)nlist, 1);
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
  for (int ln = 0; ln < nlist; ln++) {
    /* build only lists that are probed by someone: cheap enough to do all */
    oracle_pct1_b_table(d, M, ksub, centroids + (size_t)ln * d, codebooks,
                        btabs + (size_t)ln * M * ksub);
    bdone[ln] = 1;
  }
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
  for (int qi = 0; qi < nq; qi++) {
    const float *q = queries + (size_t)qi * d;
    float *atab = (float *)malloc((size_t)M * ksub * 4);
    uint64_t *heap = (uint64_t *)malloc((size_t)k * 8);
    int size = 0;
    oracle_pct1_a_table(d, M, ksub, q, codebooks, atab);
    for (int p = 0; p < nprobe; p++) {
      int64_t ln = probes[(size_t)qi * nprobe + p];
      if (ln < 0 || ln >= nlist) continue;
      int64_t beg = list_offsets[ln], end = list_offsets[ln + 1];
      if (end <= beg) continue;
      float dis0 = probe_dists[(size_t)qi * nprobe + p];
      const float *btab = btabs + (size_t)ln * M * ksub;
      /* S-term grouping (the engine's formulation, kernels.h
       * pq_sterm): the list half of T = A + B is summed per code
       * first — S = sum_m B[m][code_m], plain adds in m order,
       * exactly the engine's encode-time k_pq_sterm — then
       * dis = dis0 + S + sum_m A[m][code_m] (plain adds, m order,
       * the h:945-948 loop over A instead of T). Algebraically the
       * same T = A+B sum, grouped per vector; fp32-rounding-only
       * difference vs the per-element A+B add. */
      for (int64_t j = beg; j < end; j++) {
        uint64_t id = (uint64_t)ids[j];
        if (id & DEL_MASK) continue;               /* h:930 */
        if (bitmap_test(del_bitmap, id)) continue; /* h:935 IsValid */
        const uint8_t *code = codes + (size_t)j * M;
        float sv = 0.0f;
        for (int m = 0; m < M; m++)
          sv += btab[(size_t)m * ksub + code[m]];
        float dis = dis0 + sv;
        for (int m = 0; m < M; m++) /* h:945-948 over A */
          dis += atab[(size_t)m * ksub + code[m]];
        heap_push_or_replace(heap, &size, k,
                             make_key(dis, (uint32_t)id, 0));
      }
    }
    emit_sorted(heap, size, k, 0, out_dists + (size_t)qi * k,
                out_ids + (size_t)qi * k);
    free(atab);
    free(heap);
  }
  free(btabs);
  free(bdone);
}

/* ---- IVFFLAT search given assignments (gamma_index_ivfflat.h:36-91) ----
 * list vectors stored as fp32 d-dim codes. */
EXPORT void oracle_ivfflat_search(int nq, int d, int nlist,
                                  const float *queries,
                                  const int64_t *list_offsets,
                                  const int64_t *ids, const float *vecs,
                                  int nprobe, const int64_t *probes,
                                  const uint8_t *del_bitmap, int metric_ip,
                                  int k, float *out_dists,
                                  int64_t *out_ids) {
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
  for (int qi = 0; qi < nq; qi++) {
    const float *q = queries + (size_t)qi * d;
    uint64_t *heap = (uint64_t *)malloc((size_t)k * 8);
    int size = 0;
    for (int p = 0; p < nprobe; p++) {
      int64_t ln = probes[(size_t)qi * nprobe + p];
      if (ln < 0 || ln >= nlist) continue;
      for (int64_t j = list_offsets[ln]; j < list_offsets[ln + 1]; j++) {
        uint64_t id = (uint64_t)ids[j];
        if (id & DEL_MASK) continue;
        if (bitmap_test(del_bitmap, id)) continue;
        float dis = metric_ip ? oracle_ip(q, vecs + (size_t)j * d, d)
                              : oracle_l2sqr(q, vecs + (size_t)j * d, d);
        heap_push_or_replace(heap, &size, k,
                             make_key(dis, (uint32_t)id, metric_ip));
      }
    }
    emit_sorted(heap, size, k, metric_ip,
                out_dists + (size_t)qi * k, out_ids + (size_t)qi * k);
    free(heap);
  }
}

/* ---- coarse assign: top-nprobe centroids per query, canonical order ----
 * (quantizer->search, faiss IndexFlat equivalent; ivfpq.cc:595) */
EXPORT void oracle_coarse_assign(int nq, int d, int nlist,
                                 const float *queries,
                                 const float *centroids, int nprobe,
                                 int metric_ip, float *out_dists,
                                 int64_t *out_lists) {
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
  for (int qi = 0; qi < nq; qi++) {
    const float *q = queries + (size_t)qi * d;
    uint64_t *heap = (uint64_t *)malloc((size_t)nprobe * 8);
    int size = 0;
    for (int c = 0; c < nlist; c++) {
      float dis = metric_ip ? oracle_ip(q, centroids + (size_t)c * d, d)
                            : oracle_l2sqr(q, centroids + (size_t)c * d, d);
      heap_push_or_replace(heap, &size, nprobe,
                           make_key(dis, (uint32_t)c, metric_ip));
    }
    emit_sorted(heap, size, nprobe, metric_ip,
                out_dists + (size_t)qi * nprobe,
                out_lists + (size_t)qi * nprobe);
    free(heap);
  }
}

/* ---- PQ encode given model (pq.compute_codes, ivfpq.cc:494):
 * per subvector, nearest codeword by L2 (ties -> lowest index). ---- */
EXPORT void oracle_pq_encode(int64_t n, int d, int M, int ksub,
                             const float *residuals, const float *codebooks,
                             uint8_t *out_codes) {
  int dsub = d / M;
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (int64_t i = 0; i < n; i++) {
    const float *x = residuals + (size_t)i * d;
    for (int m = 0; m < M; m++) {
      const float *xm = x + m * dsub;
      float best = INFINITY;
      int bestj = 0;
      for (int j = 0; j < ksub; j++) {
        const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
        float acc = 0.0f;
        for (int t = 0; t < dsub; t++) {
          float diff = xm[t] - cw[t];
          acc = fmaf(diff, diff, acc);
        }
        if (acc < best) { best = acc; bestj = j; }
      }
      out_codes[(size_t)i * M + m] = (uint8_t)bestj;
    }
  }
}

EXPORT int oracle_num_threads(void) {
#ifdef _OPENMP
  return omp_get_max_threads();
#else
  return 1;
#endif
}
