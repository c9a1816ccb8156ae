/*
 * kernels.hip — hand-written gfx950 (CDNA4) kernels for the Gamma hot
 * path. No CUDA shims, no hipify output: wave64, MFMA, LDS-staged tables.
 *
 * Parity contract (DESIGN.md "Parity model"): every accumulation whose
 * value is returned to the caller uses the same sequential fmaf/add order
 * as oracle/ref_scan.c. Approximate orders (MFMA GEMM, shuffle-tree
 * reductions) are only used for candidate *selection* and are always
 * followed by the canonical re-rank.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdlib.h>

#include "kernels.h"
#include "select.hpp"

typedef float f32x4 __attribute__((ext_vector_type(4)));

#define WG 256  /* 4 waves of 64 */

/* ------------------------------------------------------------------ norms */
__global__ void k_row_norms(const float *__restrict__ v, int64_t n, int d,
                            float *__restrict__ norms) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float *row = v + i * d;
  float acc = 0.0f;
  for (int j = 0; j < d; j++) acc = fmaf(row[j], row[j], acc);
  norms[i] = acc;
}

hipError_t gk::row_norms(hipStream_t s, const float *v, int64_t n, int d,
                         float *norms) {
  if (n == 0) return hipSuccess;
  int64_t blocks = (n + WG - 1) / WG;
  k_row_norms<<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(v, n, d, norms);
  return hipGetLastError();
}

/* ------------------------------------------------------- MFMA dot GEMM
 * D[i][j] = dot(Q_i, B_j) on v_mfma_f32_16x16x4_f32 (exact f32 chain,
 * 155 TF ceiling — guide §3). One workgroup = 4 waves stacked over 64
 * query rows x 16 base columns; k-loop over d in steps of 4.
 * Lane maps (guide §3): A[i=l&15][k=l>>4], B[k=l>>4][j=l&15];
 * D: col=l&15, row=(l>>4)*4+reg. */
__global__ void __launch_bounds__(WG)
k_dots_mfma(const float *__restrict__ Q, int nq,
            const float *__restrict__ B, int64_t n, int d,
            float *__restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t col0 = (int64_t)blockIdx.x * 16;
  const int row0 = blockIdx.y * 64 + wave * 16;

  const int a_row = row0 + (lane & 15);     /* query row this lane loads */
  const int64_t b_row = col0 + (lane & 15); /* base row this lane loads */
  const int kk0 = lane >> 4;                /* k sub-index 0..3 */
  const bool a_ok = a_row < nq;
  const bool b_ok = b_row < n;
  const float *qa = Q + (int64_t)(a_ok ? a_row : 0) * d;
  const float *bb = B + (b_ok ? b_row : 0) * d;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int kk = kk0; kk < d; kk += 4) {
    float a = a_ok ? qa[kk] : 0.0f;
    float b = b_ok ? bb[kk] : 0.0f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  const int64_t col = col0 + (lane & 15);
  if (col >= n) return;
#pragma unroll
  for (int r = 0; r < 4; r++) {
    int qi = row0 + (lane >> 4) * 4 + r;
    if (qi < nq) out[(int64_t)qi * n + col] = acc[r];
  }
}

/* LDS-tiled variant: 128x128 output tile per 4-wave workgroup, BK=32,
 * +1-float row pad for conflict-free column-slice ds_reads (guide §2
 * standard fix). Each wave owns a 64x64 sub-tile = 4x4 fragments of
 * 16x16. The MFMA accumulation stays a k-ordered f32 fmaf chain (k tiles
 * processed in order; zero-padded tails add exact 0s), so the dot values
 * are bit-identical to the simple kernel and the oracle's sequential
 * chain. */
#define GT_BM 128
#define GT_BK 32
__global__ void __launch_bounds__(WG)
k_dots_mfma_tiled(const float *__restrict__ Q, int nq,
                  const float *__restrict__ B, int64_t n, int d,
                  float *__restrict__ out) {
  __shared__ float As[GT_BM][GT_BK + 1];
  __shared__ float Bs[GT_BM][GT_BK + 1];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64; /* wave row offset in tile: 0/64 */
  const int wc = (wave & 1) * 64;  /* wave col offset in tile: 0/64 */
  const int row0 = blockIdx.y * GT_BM;
  const int64_t col0 = (int64_t)blockIdx.x * GT_BM;

  f32x4 acc[4][4] = {};
  /* each thread stages 16 floats of A and B per K tile: 4 rows x float4 */
  const int ld_row = threadIdx.x >> 3;        /* 0..31 */
  const int ld_col = (threadIdx.x & 7) * 4;   /* 0,4,..,28 */

  for (int k0 = 0; k0 < d; k0 += GT_BK) {
#pragma unroll
    for (int rr = 0; rr < 4; rr++) {
      int r = ld_row + rr * 32;
      int a_row = row0 + r;
      int64_t b_row = col0 + r;
      int kk = k0 + ld_col;
      /* d % 4 == 0, so a 4-group never straddles the d boundary */
      float4 av = make_float4(0.f, 0.f, 0.f, 0.f), bv = av;
      if (kk < d) {
        if (a_row < nq)
          av = *(const float4 *)(Q + (int64_t)a_row * d + kk);
        if (b_row < n) bv = *(const float4 *)(B + b_row * d + kk);
      }
      As[r][ld_col + 0] = av.x;
      As[r][ld_col + 1] = av.y;
      As[r][ld_col + 2] = av.z;
      As[r][ld_col + 3] = av.w;
      Bs[r][ld_col + 0] = bv.x;
      Bs[r][ld_col + 1] = bv.y;
      Bs[r][ld_col + 2] = bv.z;
      Bs[r][ld_col + 3] = bv.w;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < GT_BK; kk += 4) {
      float a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; f++) {
        a[f] = As[wr + f * 16 + (lane & 15)][kk + (lane >> 4)];
        b[f] = Bs[wc + f * 16 + (lane & 15)][kk + (lane >> 4)];
      }
#pragma unroll
      for (int fi = 0; fi < 4; fi++)
#pragma unroll
        for (int fj = 0; fj < 4; fj++)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a[fi], b[fj], acc[fi][fj], 0, 0, 0);
    }
    __syncthreads();
  }
  /* D mapping: col=lane&15, row=(lane>>4)*4+r (guide §3) */
#pragma unroll
  for (int fi = 0; fi < 4; fi++) {
#pragma unroll
    for (int fj = 0; fj < 4; fj++) {
      int64_t col = col0 + wc + fj * 16 + (lane & 15);
      if (col >= n) continue;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        int qi = row0 + wr + fi * 16 + (lane >> 4) * 4 + r;
        if (qi < nq) out[(int64_t)qi * n + col] = acc[fi][fj][r];
      }
    }
  }
}

hipError_t gk::dots_mfma(hipStream_t s, const float *Q, int nq,
                         const float *B, int64_t n, int d, float *out) {
  if (nq == 0 || n == 0) return hipSuccess;
  if (nq >= 64 && n >= 64) {
    dim3 grid((uint32_t)((n + GT_BM - 1) / GT_BM),
              (uint32_t)((nq + GT_BM - 1) / GT_BM));
    k_dots_mfma_tiled<<<grid, dim3(WG), 0, s>>>(Q, nq, B, n, d, out);
  } else {
    dim3 grid((uint32_t)((n + 15) / 16), (uint32_t)((nq + 63) / 64));
    k_dots_mfma<<<grid, dim3(WG), 0, s>>>(Q, nq, B, n, d, out);
  }
  return hipGetLastError();
}

/* -------------------------------------------------- select from dist rows */
template <bool IP>
__global__ void __launch_bounds__(WG)
k_select_from_dots(int nq, int64_t ncols, int64_t col_base, int64_t ld,
                   const float *__restrict__ dots,
                   const float *__restrict__ qnorms,
                   const float *__restrict__ bnorms, int l2,
                   const uint32_t *__restrict__ bitmap, int k2,
                   uint64_t *__restrict__ state_keys, int seeded) {
  extern __shared__ char smem[];
  uint64_t *sortbuf = (uint64_t *)smem;
  uint64_t *res = sortbuf + GAMMA_SORT_CAP;
  int *state = (int *)(res + k2);

  const int q = blockIdx.x;
  if (q >= nq) return;
  const float *row = dots + (int64_t)q * ld;
  const float qn = l2 ? qnorms[q] : 0.0f;

  GammaSelector sel;
  sel.init(sortbuf, res, state, k2);
  if (seeded) sel.seed(state_keys + (int64_t)q * k2, k2);

  const int64_t chunk = (int64_t)blockDim.x * 2;
  for (int64_t c0 = 0; c0 < ncols; c0 += chunk) {
    int64_t cend = min(c0 + chunk, ncols);
    for (int64_t c = c0 + threadIdx.x; c < cend; c += blockDim.x) {
      int64_t id = col_base + c;
      if (gamma_bitmap_test(bitmap, (uint64_t)id)) continue;
      float dot = row[c];
      float dist = l2 ? fmaf(-2.0f, dot, qn + bnorms[id]) : dot;
      sel.push(gamma_make_key<IP>(dist, (uint32_t)id));
    }
    sel.maybe_flush(2 * blockDim.x);
  }
  sel.finish();
  for (int i = threadIdx.x; i < k2; i += blockDim.x)
    state_keys[(int64_t)q * k2 + i] = res[i];
}

/* Wave-per-query variant for small k2 (coarse assign: k2 = nprobe).
 * One block-wide selector per 4096-float row is barrier-dominated (it
 * measured 490 us/step, reading 164 MB at 334 GB/s); here each of the
 * 8 waves in a WG owns one query with a private GammaWaveSelector, so
 * the scan has no block barriers at all and flushes are rare (expected
 * pushes per wave ~ k2 * ln(ncols/k2)). Exact: same key order as the
 * block path (both validated against each other in tests). */
template <bool IP>
__global__ void __launch_bounds__(512)
k_select_from_dots_wave(int nq, int64_t ncols, int64_t col_base,
                        int64_t ld, const float *__restrict__ dots,
                        const float *__restrict__ qnorms,
                        const float *__restrict__ bnorms, int l2,
                        const uint32_t *__restrict__ bitmap, int k2,
                        uint64_t *__restrict__ state_keys, int seeded) {
  extern __shared__ char smem[];
  const int nw = 512 / 64;
  uint64_t *wb = (uint64_t *)smem;
  int *cnts = (int *)(wb + (size_t)nw * GAMMA_WSEL_CAP);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int q = blockIdx.x * nw + wave;
  if (q >= nq) return; /* q is wave-uniform: the whole wave exits */
  const float *row = dots + (int64_t)q * ld;
  const float qn = l2 ? qnorms[q] : 0.0f;

  GammaWaveSelector sel;
  sel.init(wb + (size_t)wave * GAMMA_WSEL_CAP, cnts + wave, k2);
  if (seeded) { /* fold previous segments' keys in (k2 <= cap here) */
    for (int i = lane; i < k2; i += 64) {
      uint64_t kk = state_keys[(int64_t)q * k2 + i];
      if (kk != GAMMA_KEY_EMPTY) {
        int idx = atomicAdd(sel.cnt, 1);
        sel.buf[sel.k + idx] = kk;
      }
    }
    sel.finish();
  }
  if ((ncols & 3) == 0 && (col_base & 3) == 0 && ((uintptr_t)row & 15) == 0) {
    /* vectorized: 4 dists (and 4 bnorms) per lane per pass — quarters
     * the serial trip count over a 16k-wide row (nlist=16384 coarse
     * assign reads 64 KB/query; the scalar loop was latency-bound) */
    const float4 *row4 = (const float4 *)row;
    for (int64_t c0 = 0; c0 < (ncols >> 2); c0 += 64) {
      int64_t c4 = c0 + lane;
      if (c4 < (ncols >> 2)) {
        int64_t id0 = col_base + c4 * 4;
        float4 d4 = row4[c4];
        float4 b4 = l2 ? *(const float4 *)(bnorms + id0)
                       : make_float4(0, 0, 0, 0);
        float dv[4] = {d4.x, d4.y, d4.z, d4.w};
        float bv[4] = {b4.x, b4.y, b4.z, b4.w};
#pragma unroll
        for (int t = 0; t < 4; t++) {
          int64_t id = id0 + t;
          if (!gamma_bitmap_test(bitmap, (uint64_t)id)) {
            float dist = l2 ? fmaf(-2.0f, dv[t], qn + bv[t]) : dv[t];
            sel.push(gamma_make_key<IP>(dist, (uint32_t)id));
          }
        }
      }
      sel.maybe_flush(64 * 4);
    }
  } else {
    for (int64_t c0 = 0; c0 < ncols; c0 += 64) {
      int64_t c = c0 + lane;
      if (c < ncols) {
        int64_t id = col_base + c;
        if (!gamma_bitmap_test(bitmap, (uint64_t)id)) {
          float dot = row[c];
          float dist = l2 ? fmaf(-2.0f, dot, qn + bnorms[id]) : dot;
          sel.push(gamma_make_key<IP>(dist, (uint32_t)id));
        }
      }
      sel.maybe_flush(64);
    }
  }
  sel.finish();
  for (int i = lane; i < k2; i += 64)
    state_keys[(int64_t)q * k2 + i] = sel.buf[i];
}

hipError_t gk::select_from_dots(hipStream_t s, int nq, int64_t ncols,
                                int64_t col_base, int64_t ld,
                                const float *dots, const float *qnorms,
                                const float *bnorms, bool l2, bool ip_order,
                                const uint32_t *bitmap, int k2,
                                uint64_t *state_keys, bool seeded) {
  /* k2 <= 192: seeds (k2 keys) plus one 64-push interval must fit the
   * wave cap (GAMMA_WSEL_CAP - k2), with slack; covers every coarse
   * assign (nprobe). Larger k2 (FLAT top-k accumulation) -> block path. */
  if (k2 <= 192) {
    const int nw = 512 / 64;
    size_t smem = (size_t)nw * GAMMA_WSEL_CAP * 8 + nw * sizeof(int);
    dim3 g((uint32_t)((nq + nw - 1) / nw));
    if (ip_order)
      k_select_from_dots_wave<true><<<g, dim3(512), smem, s>>>(
          nq, ncols, col_base, ld, dots, qnorms, bnorms, l2 ? 1 : 0,
          bitmap, k2, state_keys, seeded ? 1 : 0);
    else
      k_select_from_dots_wave<false><<<g, dim3(512), smem, s>>>(
          nq, ncols, col_base, ld, dots, qnorms, bnorms, l2 ? 1 : 0,
          bitmap, k2, state_keys, seeded ? 1 : 0);
    return hipGetLastError();
  }
  size_t smem = (GAMMA_SORT_CAP + k2) * 8 + 4 * sizeof(int);
  if (ip_order)
    k_select_from_dots<true><<<dim3(nq), dim3(WG), smem, s>>>(
        nq, ncols, col_base, ld, dots, qnorms, bnorms, l2 ? 1 : 0, bitmap,
        k2, state_keys, seeded ? 1 : 0);
  else
    k_select_from_dots<false><<<dim3(nq), dim3(WG), smem, s>>>(
        nq, ncols, col_base, ld, dots, qnorms, bnorms, l2 ? 1 : 0, bitmap,
        k2, state_keys, seeded ? 1 : 0);
  return hipGetLastError();
}

/* ----------------------------------------------- full-sort row select */
template <bool IP>
__global__ void __launch_bounds__(512)
k_select_rows_full(int nq, int ncols, int64_t ld,
                   const float *__restrict__ dots,
                   const float *__restrict__ qnorms,
                   const float *__restrict__ bnorms, int l2, int k2,
                   float *__restrict__ out_dists,
                   int64_t *__restrict__ out_ids) {
  extern __shared__ char smem[];
  uint64_t *buf = (uint64_t *)smem;
  const int q = blockIdx.x;
  if (q >= nq) return;
  int n = 1;
  while (n < ncols || n < k2) n <<= 1;
  const float *row = dots + (int64_t)q * ld;
  const float qn = l2 ? qnorms[q] : 0.0f;
  for (int c = threadIdx.x; c < n; c += blockDim.x) {
    uint64_t key = GAMMA_KEY_EMPTY;
    if (c < ncols) {
      float dot = row[c];
      float dist = l2 ? fmaf(-2.0f, dot, qn + bnorms[c]) : dot;
      key = gamma_make_key<IP>(dist, (uint32_t)c);
    }
    buf[c] = key;
  }
  gamma_bitonic_sort(buf, n);
  for (int i = threadIdx.x; i < k2; i += blockDim.x) {
    uint64_t key = buf[i];
    out_dists[(int64_t)q * k2 + i] =
        (key == GAMMA_KEY_EMPTY) ? -1.0f : gamma_key_dist<IP>(key);
    out_ids[(int64_t)q * k2 + i] = gamma_key_id(key);
  }
}

hipError_t gk::select_rows_full(hipStream_t s, int nq, int ncols,
                                int64_t ld, const float *dots,
                                const float *qnorms, const float *bnorms,
                                bool l2, bool ip_order, int k2,
                                float *out_dists, int64_t *out_ids) {
  int n = 1;
  while (n < ncols || n < k2) n <<= 1;
  size_t smem = (size_t)n * 8;
  if (smem > 64 * 1024) return hipErrorInvalidValue;
  if (ip_order)
    k_select_rows_full<true><<<dim3(nq), dim3(512), smem, s>>>(
        nq, ncols, ld, dots, qnorms, bnorms, l2 ? 1 : 0, k2, out_dists,
        out_ids);
  else
    k_select_rows_full<false><<<dim3(nq), dim3(512), smem, s>>>(
        nq, ncols, ld, dots, qnorms, bnorms, l2 ? 1 : 0, k2, out_dists,
        out_ids);
  return hipGetLastError();
}

/* --------------------------------------------------------------- argmin */
__global__ void k_argmin_rows(int64_t nrows, int ncols,
                              const float *__restrict__ dots,
                              const float *__restrict__ qnorms,
                              const float *__restrict__ bnorms, int l2,
                              int32_t *__restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nrows) return;
  const float *row = dots + i * ncols;
  float qn = l2 ? qnorms[i] : 0.0f;
  float best = INFINITY;
  int bestj = 0;
  for (int j = 0; j < ncols; j++) {
    float v = l2 ? fmaf(-2.0f, row[j], qn + bnorms[j]) : -row[j];
    if (v < best) { best = v; bestj = j; }
  }
  out[i] = bestj;
}

hipError_t gk::argmin_rows(hipStream_t s, int64_t nrows, int ncols,
                           const float *dots, const float *qnorms,
                           const float *bnorms, bool l2, int32_t *out) {
  if (nrows == 0) return hipSuccess;
  int64_t blocks = (nrows + WG - 1) / WG;
  k_argmin_rows<<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(
      nrows, ncols, dots, qnorms, bnorms, l2 ? 1 : 0, out);
  return hipGetLastError();
}

/* ------------------------------------------------------- IVFPQ fused scan
 * One workgroup per query.
 * L2: per probed list stage T = A_q + B_list in LDS (the decomposed
 * use_precomputed_table=1 tables, ivfpq.h:254-262) with dis0 = the
 * coarse probe distance; IP: one query-level table (h:164-167) with
 * dis0 = dot(q, c_list) in canonical order.
 * Scan (h:923-953): skip delete-marked ids and bitmap-deleted docs,
 * dis = dis0 + sum_m T[m][code_m] in ascending m (plain adds — matches
 * oracle bit-for-bit, see oracle_ivfpq_search_pct1).
 * MW = M/4 compile-time (0 = generic runtime-M path). The templated path
 * stages GAMMA_ADC_C codes per thread in registers with all global loads
 * issued before any use. C=1 measured fastest at BS=512 (9.06 ms vs
 * 9.84 ms for C=2 on the uniform 10M/nprobe=32 microbench; 24 waves/CU
 * already cover the load latency, and C=1 halves the register-staging
 * pressure) — see tools/adc_bench.hip. */
#define GAMMA_ADC_C_DEFAULT 1
template <bool IP, int MW, int BS, int CP>
__global__ void __launch_bounds__(BS)
k_ivfpq_scan(int nq, int S, int d, int M, int nprobe, int k2,
             const float *__restrict__ queries,
             const float *__restrict__ centroids,
             const float *__restrict__ codebooks,
             const float *__restrict__ atab,
             const float *__restrict__ probe_dists,
             const GammaBucketDev *__restrict__ buckets, int nlist,
             const int64_t *__restrict__ probes,
             const uint32_t *__restrict__ bitmap,
             uint64_t *__restrict__ out_keys,
             const int *__restrict__ kill_flag,
             const int32_t *__restrict__ qmap) {
  extern __shared__ char smem[];
  const int ksub = 256;
  const int dsub = d / M;
  float *lut = (float *)smem;                       /* M*ksub */
  uint64_t *sortbuf = (uint64_t *)(smem + ((size_t)M * ksub * 4 + 7) / 8 * 8);
  uint64_t *res = sortbuf + GAMMA_SORT_CAP;
  float *qs = (float *)(res + k2);                  /* d (IP table build) */
  float *dis0s = qs + d;                            /* 2 (one per half) */
  long long *szsh = (long long *)(dis0s + 2);       /* 2 (list sizes) */
  int *state = (int *)(szsh + 2) + 1;               /* int[2] */

  /* probe-split: S sub-workgroups per query, sub-block handles probes
   * p ≡ sub (mod S); partials merged by sort_rows afterwards. S>1 is
   * the small-batch/serving path (a lone query still fills S CUs). */
  const int bq = blockIdx.x / S;
  const int sub = blockIdx.x - bq * S;
  if (bq >= nq) return;
  const int q = qmap ? qmap[bq] : bq; /* scheduled query (cache
                                         clustering); outputs go to the
                                         ORIGINAL row below */
  const float *qg = queries + (int64_t)q * d;
  for (int i = threadIdx.x; i < d; i += blockDim.x) qs[i] = qg[i];

  GammaSelector sel;
  sel.init(sortbuf, res, state, k2);  /* has the barrier qs needs */

  if (IP) {
    /* query-level table T[m][j] = dot(q_m, cw_mj), canonical per entry */
    for (int e = threadIdx.x; e < M * ksub; e += blockDim.x) {
      int m = e >> 8, j = e & 255;
      const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
      const float *qm = qs + m * dsub;
      float acc = 0.0f;
      for (int t = 0; t < dsub; t++) acc = fmaf(qm[t], cw[t], acc);
      lut[e] = acc;
    }
    __syncthreads();
  } else {
    /* L2: the query-level A table ONCE per workgroup. The list half of
     * the decomposition (B) is pre-folded into one float per vector
     * (bucket svals, gk::pq_sterm), so no per-(query,list) table is
     * ever staged: dis = coarse_dis + S_v + sum_m A[m][code_m]. */
    const float4 *Aq = (const float4 *)(atab + (size_t)q * M * ksub);
    float4 *lut4 = (float4 *)lut;
    for (int e = threadIdx.x; e < (M * ksub) >> 2; e += blockDim.x)
      lut4[e] = Aq[e];
    __syncthreads();
  }

  /* in-flight kill (is_killed_every<1024> analog, ivfpq.h:927): poll a
   * device flag between lists with an agent-scope load (L2-served, so a
   * host write during the kernel is visible — plain loads can stay
   * L1-stale, microarch §Workgroup dispatch). */
  if (MW > 0) {
    /* TWO probed lists in flight per workgroup: halves of the block
     * scan consecutive probes of this sub-workgroup's sequence. At
     * large nlist a list holds ~N/nlist codes (~600 at the headline
     * config) — against a 512-thread block that wastes ~30% of the
     * last pass; 2x(BS/2) halves the tail waste and doubles the
     * independent HBM streams. Push order changes; the selector's
     * top-k2 is exact under the (dist,id) total order, so results are
     * unchanged. */
    const int C = CP;
    const int HB = BS / 2;
    const int half = threadIdx.x >= HB ? 1 : 0;
    const int tid = threadIdx.x - half * HB;
    for (int t = 0;; t += 2) {
      int p0 = sub + t * S;
      if (p0 >= nprobe) break;
      if (kill_flag &&
          __hip_atomic_load(kill_flag, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_AGENT))
        break;
      int p = sub + (t + half) * S;
      int64_t ln = -1;
      if (p < nprobe) ln = probes[(int64_t)q * nprobe + p];
      if (ln >= nlist) ln = -1;
      GammaBucketDev bk;
      bk.size = 0;
      if (ln >= 0) bk = buckets[ln];
      float dis0 = 0.0f;
      if (IP) {
        if (tid == 0) { /* dis0 = dot(q, c), canonical order, per half */
          float acc = 0.0f;
          if (ln >= 0) {
            const float *cent = centroids + (size_t)ln * d;
            for (int e = 0; e < d; e++) acc = fmaf(qs[e], cent[e], acc);
          }
          dis0s[half] = acc;
          szsh[half] = bk.size;
        }
      } else {
        /* dis0 = the coarse probe distance (ivfpq.h:255
         * dis0=coarse_dis); the per-list table half arrives as the
         * per-vector S term */
        if (ln >= 0 && p < nprobe)
          dis0 = probe_dists[(int64_t)q * nprobe + p];
        if (tid == 0) szsh[half] = bk.size;
      }
      __syncthreads();
      if (IP) dis0 = dis0s[half];
      /* both halves iterate in lockstep to the longer list so the
       * selector's block-wide flush barriers stay aligned */
      long long maxsz = szsh[0] > szsh[1] ? szsh[0] : szsh[1];
      const uint32_t *ids = bk.ids;
      const uint8_t *codes = (const uint8_t *)bk.data;
      const float *svals = (const float *)bk.svals;
      for (long long j0 = 0; j0 < maxsz; j0 += (long long)HB * C) {
        long long jb = j0 + (long long)tid * C;
        uint32_t w[C][MW ? MW : 1]; /* compile-time bounds -> registers */
        int64_t idv[C];
        float sv[C];
#pragma unroll
        for (int c = 0; c < C; c++) {
          long long j = jb + c;
          if (ln >= 0 && j < bk.size) {
            idv[c] = (int64_t)(int32_t)ids[j]; /* bit31 -> negative */
            if (!IP) sv[c] = svals[j];
            const uint32_t *cw = (const uint32_t *)(codes + (size_t)j * M);
#pragma unroll
            for (int mw = 0; mw < MW; mw++) w[c][mw] = cw[mw];
          } else {
            idv[c] = -1; /* bit 63 set -> skipped below */
            sv[c] = 0.0f;
          }
        }
#pragma unroll
        for (int c = 0; c < C; c++) {
          int64_t id = idv[c];
          if (!((uint64_t)id >> 63) &&
              !gamma_bitmap_test(bitmap, (uint64_t)id)) {
            float dis = IP ? dis0 : dis0 + sv[c];
            const float *tab = lut;
#pragma unroll
            for (int mw = 0; mw < MW; mw++) {
              uint32_t wv = w[c][mw];
              dis += tab[wv & 255u];         tab += ksub;
              dis += tab[(wv >> 8) & 255u];  tab += ksub;
              dis += tab[(wv >> 16) & 255u]; tab += ksub;
              dis += tab[wv >> 24];          tab += ksub;
            }
            sel.push(gamma_make_key<IP>(dis, (uint32_t)id));
          }
        }
        sel.maybe_flush(blockDim.x * C);
      }
      __syncthreads(); /* dis0s/szsh rewritten next pair */
    }
  } else {
    for (int p = sub; p < nprobe; p += S) {
      if (kill_flag &&
          __hip_atomic_load(kill_flag, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_AGENT))
        break;
      int64_t ln = probes[(int64_t)q * nprobe + p];
      if (ln < 0 || ln >= nlist) continue;
      GammaBucketDev bk = buckets[ln];
      if (bk.size <= 0) continue;
      const float *cent = centroids + (size_t)ln * d;

      float dis0;
      if (IP) {
        if (threadIdx.x == 0) {  /* dis0 = dot(q, c), canonical order */
          float acc = 0.0f;
          for (int e = 0; e < d; e++) acc = fmaf(qs[e], cent[e], acc);
          dis0s[0] = acc;
        }
        __syncthreads();
        dis0 = dis0s[0];
      } else {
        dis0 = probe_dists[(int64_t)q * nprobe + p];
      }

      const uint32_t *ids = bk.ids;
      const uint8_t *codes = (const uint8_t *)bk.data;
      const float *svals = (const float *)bk.svals;
      const int mwords = M >> 2;
      for (long long j0 = 0; j0 < bk.size; j0 += blockDim.x) {
        long long j = j0 + threadIdx.x;
        if (j < bk.size) {
          int64_t id = (int64_t)(int32_t)ids[j];
          if (!((uint64_t)id >> 63) &&
              !gamma_bitmap_test(bitmap, (uint64_t)id)) {
            const uint32_t *cw = (const uint32_t *)(codes + (size_t)j * M);
            float dis = IP ? dis0 : dis0 + svals[j];
            const float *tab = lut;
            for (int mw = 0; mw < mwords; mw++) {
              uint32_t wv = cw[mw];
              dis += tab[wv & 255u];         tab += ksub;
              dis += tab[(wv >> 8) & 255u];  tab += ksub;
              dis += tab[(wv >> 16) & 255u]; tab += ksub;
              dis += tab[wv >> 24];          tab += ksub;
            }
            sel.push(gamma_make_key<IP>(dis, (uint32_t)id));
          }
        }
        sel.maybe_flush(blockDim.x);
      }
      if (IP) __syncthreads(); /* dis0s rewritten next list */
    }
  }
  sel.finish();
  for (int i = threadIdx.x; i < k2; i += blockDim.x)
    out_keys[((int64_t)q * S + sub) * k2 + i] = res[i];
}

__global__ void k_extract_probe0(int nq, int nprobe,
                                 const int64_t *__restrict__ probes,
                                 int32_t *__restrict__ out) {
  int q = blockIdx.x * blockDim.x + threadIdx.x;
  if (q < nq) out[q] = (int32_t)probes[(int64_t)q * nprobe];
}

hipError_t gk::extract_probe0(hipStream_t s, int nq, int nprobe,
                              const int64_t *probes, int32_t *out) {
  k_extract_probe0<<<dim3((uint32_t)((nq + WG - 1) / WG)), dim3(WG), 0,
                     s>>>(nq, nprobe, probes, out);
  return hipGetLastError();
}

hipError_t gk::ivfpq_scan(hipStream_t s, int nq, int S, int d, int M,
                          int nprobe,
                          int k2, const float *queries,
                          const float *centroids, const float *codebooks,
                          const float *atab,
                          const float *probe_dists,
                          const GammaBucketDev *buckets, int nlist,
                          const int64_t *probes, const uint32_t *bitmap,
                          bool ip, uint64_t *out_keys,
                          const int *kill_flag, const int32_t *qmap) {
  size_t smem = ((size_t)M * 256 * 4 + 7) / 8 * 8 +
                (GAMMA_SORT_CAP + k2) * 8 + (d + 2) * 4 +
                2 * sizeof(long long) + 4 * sizeof(int);
  if (smem > 160 * 1024) return hipErrorInvalidValue;
  /* batched path needs flush margin blockDim*C inside the selector cap.
   * 512-thread blocks put 24 waves on a CU at the same LDS/WG (the ADC
   * gather stream wants ~4 waves/SIMD — microarch §LDS). */
  /* BS=256 wins when lists are short (nlist=16384: ~N/nlist codes per
   * list barely fills 512 threads; 256 halves the tail waste and puts
   * more independent workgroups on each CU). GAMMA_SCAN_BS overrides
   * for experiments. */
  int BS = 512;
  {
    const char *e = getenv("GAMMA_SCAN_BS");
    if (e && (atoi(e) == 256 || atoi(e) == 512)) BS = atoi(e);
  }
  /* C = codes register-staged per thread per flush interval: deeper
   * staging doubles the outstanding VMEM requests per wave — the lever
   * when short lists (large nlist) cap the per-wave pipelining.
   * GAMMA_ADC_C overrides; default policy below. */
  int CSEL = GAMMA_ADC_C_DEFAULT;
  {
    const char *e = getenv("GAMMA_ADC_C");
    if (e && (atoi(e) == 1 || atoi(e) == 2)) CSEL = atoi(e);
  }
  bool fast = (k2 + BS * CSEL) <= GAMMA_SORT_CAP &&
              (M == 16 || M == 32 || M == 64 || M == 96);
  if (!((k2 + BS * CSEL) <= GAMMA_SORT_CAP)) CSEL = 1;
  dim3 g((uint32_t)nq * (uint32_t)S);
#define GAMMA_LAUNCH_SCAN(IPV, MWV, BSV, CPV)                             \
  k_ivfpq_scan<IPV, MWV, BSV, CPV><<<g, dim3(BSV), smem, s>>>(            \
      nq, S, d, M, nprobe, k2, queries, centroids, codebooks, atab,       \
      probe_dists, buckets, nlist, probes, bitmap, out_keys,              \
      kill_flag, qmap)
#define GAMMA_LAUNCH_SCAN_BS(IPV, MWV)                                    \
  do {                                                                    \
    if (BS == 256 && CSEL == 2) GAMMA_LAUNCH_SCAN(IPV, MWV, 256, 2);      \
    else if (BS == 256) GAMMA_LAUNCH_SCAN(IPV, MWV, 256, 1);              \
    else if (CSEL == 2) GAMMA_LAUNCH_SCAN(IPV, MWV, 512, 2);              \
    else GAMMA_LAUNCH_SCAN(IPV, MWV, 512, 1);                             \
  } while (0)
  if (ip) {
    if (!fast) GAMMA_LAUNCH_SCAN(true, 0, WG, 1);
    else if (M == 16) GAMMA_LAUNCH_SCAN_BS(true, 4);
    else if (M == 32) GAMMA_LAUNCH_SCAN_BS(true, 8);
    else if (M == 64) GAMMA_LAUNCH_SCAN_BS(true, 16);
    else GAMMA_LAUNCH_SCAN_BS(true, 24);
  } else {
    if (!fast) GAMMA_LAUNCH_SCAN(false, 0, WG, 1);
    else if (M == 16) GAMMA_LAUNCH_SCAN_BS(false, 4);
    else if (M == 32) GAMMA_LAUNCH_SCAN_BS(false, 8);
    else if (M == 64) GAMMA_LAUNCH_SCAN_BS(false, 16);
    else GAMMA_LAUNCH_SCAN_BS(false, 24);
  }
#undef GAMMA_LAUNCH_SCAN_BS
#undef GAMMA_LAUNCH_SCAN
  return hipGetLastError();
}

/* ------------------------------------------------------ IVFFLAT fused scan
 * Lists store raw fp32 vectors (gamma_index_ivfflat.h:63-91). One wave
 * handles one list vector per pass: 64 lanes x ceil(d/64) elements each,
 * coalesced 512 B reads at d=128, shuffle-tree reduction (selection only;
 * final distances are canonicalized by the re-rank kernel). */
template <bool IP>
__global__ void __launch_bounds__(WG)
k_ivfflat_scan(int nq, int d, int nprobe, int k2,
               const float *__restrict__ queries,
               const GammaBucketDev *__restrict__ buckets, int nlist,
               const int64_t *__restrict__ probes,
               const uint32_t *__restrict__ bitmap,
               uint64_t *__restrict__ out_keys) {
  extern __shared__ char smem[];
  uint64_t *sortbuf = (uint64_t *)smem;
  uint64_t *res = sortbuf + GAMMA_SORT_CAP;
  float *qs = (float *)(res + k2);
  int *state = (int *)(qs + d);

  const int q = blockIdx.x;
  if (q >= nq) return;
  const float *qg = queries + (int64_t)q * d;
  for (int i = threadIdx.x; i < d; i += blockDim.x) qs[i] = qg[i];

  GammaSelector sel;
  sel.init(sortbuf, res, state, k2);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwave = blockDim.x >> 6;
  const int half = lane >> 5;   /* two vectors per wave: one per half */
  const int hlane = lane & 31;

  for (int p = 0; p < nprobe; p++) {
    int64_t ln = probes[(int64_t)q * nprobe + p];
    if (ln < 0 || ln >= nlist) continue;
    GammaBucketDev bk = buckets[ln];
    const uint32_t *ids = bk.ids;
    const float *vecs = (const float *)bk.data;

    const int vpp = nwave * 2; /* vectors per WG pass */
    long long iters = (bk.size + vpp - 1) / vpp;
    for (long long it = 0; it < iters; it++) {
      long long j = it * vpp + wave * 2 + half;
      bool live = false;
      uint64_t id = 0;
      float dist = 0.0f;
      if (j < bk.size) {
        id = (uint64_t)(int64_t)(int32_t)ids[j];
        live = !(id >> 63) && !gamma_bitmap_test(bitmap, id);
        if (live) {
          /* float4 per half-wave lane: 32 x 16 B = one 512-B coalesced
           * read per vector at d=128 (selection only; the canonical
           * re-rank recomputes final distances) */
          const float4 *v4 = (const float4 *)(vecs + (size_t)j * d);
          const float4 *q4 = (const float4 *)qs;
          float acc = 0.0f;
          for (int e = hlane; e < (d >> 2); e += 32) {
            float4 a = q4[e], b = v4[e];
            if (IP) {
              acc = fmaf(a.x, b.x, acc);
              acc = fmaf(a.y, b.y, acc);
              acc = fmaf(a.z, b.z, acc);
              acc = fmaf(a.w, b.w, acc);
            } else {
              float dx = a.x - b.x, dy = a.y - b.y, dz = a.z - b.z,
                    dw = a.w - b.w;
              acc = fmaf(dx, dx, acc);
              acc = fmaf(dy, dy, acc);
              acc = fmaf(dz, dz, acc);
              acc = fmaf(dw, dw, acc);
            }
          }
          for (int off = 16; off > 0; off >>= 1)
            acc += __shfl_xor(acc, off, 64); /* within each 32-half */
          dist = acc;
        }
      }
      /* rotate the pushing hlane so buffers fill evenly */
      if (live && hlane == (int)(it & 31))
        sel.push(gamma_make_key<IP>(dist, (uint32_t)id));
      if ((it & 15) == 15) sel.maybe_flush(blockDim.x);
    }
    sel.maybe_flush(blockDim.x);
  }
  sel.finish();
  for (int i = threadIdx.x; i < k2; i += blockDim.x)
    out_keys[(int64_t)q * k2 + i] = res[i];
}

hipError_t gk::ivfflat_scan(hipStream_t s, int nq, int d, int nprobe, int k2,
                            const float *queries,
                            const GammaBucketDev *buckets, int nlist,
                            const int64_t *probes, const uint32_t *bitmap,
                            bool ip, uint64_t *out_keys) {
  size_t smem = (GAMMA_SORT_CAP + k2) * 8 + d * 4 + 4 * sizeof(int);
  if (smem > 160 * 1024) return hipErrorInvalidValue;
  if (ip)
    k_ivfflat_scan<true><<<dim3(nq), dim3(WG), smem, s>>>(
        nq, d, nprobe, k2, queries, buckets, nlist, probes, bitmap,
        out_keys);
  else
    k_ivfflat_scan<false><<<dim3(nq), dim3(WG), smem, s>>>(
        nq, d, nprobe, k2, queries, buckets, nlist, probes, bitmap,
        out_keys);
  return hipGetLastError();
}

/* ------------------------------------------------------ FLAT stream scan
 * Per-query workgroup streaming every segment (small-nq path; HBM-bound).
 * Distances here are selection-only (canonical re-rank follows). */
template <bool IP>
__global__ void __launch_bounds__(WG)
k_flat_stream(int nq, int64_t n, int d, int k2,
              const float *__restrict__ queries,
              const float *const *__restrict__ segs, int seg_shift,
              const uint32_t *__restrict__ bitmap,
              uint64_t *__restrict__ out_keys) {
  extern __shared__ char smem[];
  uint64_t *sortbuf = (uint64_t *)smem;
  uint64_t *res = sortbuf + GAMMA_SORT_CAP;
  float *qs = (float *)(res + k2);
  int *state = (int *)(qs + d);

  const int q = blockIdx.x;
  if (q >= nq) return;
  const float *qg = queries + (int64_t)q * d;
  for (int i = threadIdx.x; i < d; i += blockDim.x) qs[i] = qg[i];

  GammaSelector sel;
  sel.init(sortbuf, res, state, k2);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwave = blockDim.x >> 6;
  const int64_t seg_mask = ((int64_t)1 << seg_shift) - 1;

  long long iters = (n + nwave - 1) / nwave;
  for (long long it = 0; it < iters; it++) {
    int64_t j = it * nwave + wave;
    bool live = false;
    float dist = 0.0f;
    if (j < n) {
      live = !gamma_bitmap_test(bitmap, (uint64_t)j);
      if (live) {
        const float2 *v2 = (const float2 *)(
            segs[j >> seg_shift] + (size_t)(j & seg_mask) * d);
        const float2 *q2 = (const float2 *)qs;
        float acc = 0.0f;
        for (int e = lane; e < (d >> 1); e += 64) {
          float2 a = q2[e], b = v2[e];
          if (IP) {
            acc = fmaf(a.x, b.x, acc);
            acc = fmaf(a.y, b.y, acc);
          } else {
            float dx = a.x - b.x, dy = a.y - b.y;
            acc = fmaf(dx, dx, acc);
            acc = fmaf(dy, dy, acc);
          }
        }
        for (int off = 32; off > 0; off >>= 1)
          acc += __shfl_xor(acc, off, 64);
        dist = acc;
      }
    }
    if (live && lane == (int)(j & 63))
      sel.push(gamma_make_key<IP>(dist, (uint32_t)j));
    if ((it & 15) == 15) sel.maybe_flush(blockDim.x);
  }
  sel.finish();
  for (int i = threadIdx.x; i < k2; i += blockDim.x)
    out_keys[(int64_t)q * k2 + i] = res[i];
}

hipError_t gk::flat_stream_scan(hipStream_t s, int nq, int64_t n, int d,
                                int k2, const float *queries,
                                const float *const *segs, int seg_shift,
                                const uint32_t *bitmap, bool ip,
                                uint64_t *out_keys) {
  size_t smem = (GAMMA_SORT_CAP + k2) * 8 + d * 4 + 4 * sizeof(int);
  if (smem > 160 * 1024) return hipErrorInvalidValue;
  if (ip)
    k_flat_stream<true><<<dim3(nq), dim3(WG), smem, s>>>(
        nq, n, d, k2, queries, segs, seg_shift, bitmap, out_keys);
  else
    k_flat_stream<false><<<dim3(nq), dim3(WG), smem, s>>>(
        nq, n, d, k2, queries, segs, seg_shift, bitmap, out_keys);
  return hipGetLastError();
}

/* ------------------------------------------------------------- re-rank
 * Canonical-order exact distances (matches oracle_l2sqr/oracle_ip).
 * Thread per (query, candidate); consecutive threads share the query so
 * its elements broadcast from L1. */
template <bool IP, int DV>
__global__ void k_rerank(int nq, int ncand, int d,
                         const float *__restrict__ queries,
                         const float *const *__restrict__ segs,
                         int seg_shift, const uint64_t *__restrict__ keys_in,
                         uint64_t *__restrict__ keys_out) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (int64_t)nq * ncand) return;
  int q = (int)(idx / ncand);
  uint64_t key = keys_in[idx];
  if (key == GAMMA_KEY_EMPTY) { keys_out[idx] = GAMMA_KEY_EMPTY; return; }
  uint32_t id = (uint32_t)(key & 0xffffffffu);
  const int64_t seg_mask = ((int64_t)1 << seg_shift) - 1;
  const float *v = segs[id >> seg_shift] + (size_t)(id & seg_mask) * d;
  const float *qv = queries + (int64_t)q * d;
  /* float4 loads (d % 4 == 0), scalar fmaf chain in canonical order.
   * DV != 0 makes the trip count compile-time so the loads pipeline
   * ahead of the dependent fmaf chain. */
  const float4 *v4 = (const float4 *)v;
  const float4 *q4 = (const float4 *)qv;
  float acc = 0.0f;
  const int nt4 = DV ? (DV >> 2) : (d >> 2);
#pragma unroll 8
  for (int t = 0; t < nt4; t++) {
    float4 a = q4[t], b = v4[t];
    if (IP) {
      acc = fmaf(a.x, b.x, acc);
      acc = fmaf(a.y, b.y, acc);
      acc = fmaf(a.z, b.z, acc);
      acc = fmaf(a.w, b.w, acc);
    } else {
      float dx = a.x - b.x, dy = a.y - b.y, dz = a.z - b.z,
            dw = a.w - b.w;
      acc = fmaf(dx, dx, acc);
      acc = fmaf(dy, dy, acc);
      acc = fmaf(dz, dz, acc);
      acc = fmaf(dw, dw, acc);
    }
  }
  keys_out[idx] = gamma_make_key<IP>(acc, id);
}

/* LDS-staged re-rank (tools/rerank_bench.hip "v2": 0.354 ms vs 0.698 ms
 * for the thread-per-candidate version at nq=10k ncand=200 d=128 —
 * 2.95 TB/s algorithmic). One block per (query, candidate chunk); the
 * block stages BS candidate rows through LDS in 128 B slices with
 * 8 threads cooperating per row (coalesced float4s), then each thread
 * accumulates ITS candidate's distance from the staged slice in the
 * same canonical element order as the scalar chain — bit-identical
 * output, ~4x less time stalled on scattered row gathers. */
template <bool IP, int DV, int BS>
__global__ void __launch_bounds__(BS)
k_rerank_lds(int nq, int ncand, const float *__restrict__ queries,
             const float *const *__restrict__ segs, int n_segs,
             int seg_shift, const uint64_t *__restrict__ keys_in,
             uint64_t *__restrict__ keys_out) {
  const int CH = 32; /* floats per slice = 128 B */
  const int SEG_LDS = 64; /* seg-table cache (64 segs = 33M vectors) */
  __shared__ float rows[BS][CH + 1]; /* +1: bank-shift */
  __shared__ float qs[DV];
  __shared__ const float *seg_lds[SEG_LDS];
  const int nchunk = (ncand + BS - 1) / BS;
  const int q = blockIdx.x / nchunk;
  const int c0 = (blockIdx.x - q * nchunk) * BS;
  if (q >= nq) return;
  for (int i = threadIdx.x; i < DV; i += BS)
    qs[i] = queries[(size_t)q * DV + i];
  /* the row gathers chase segs[id>>shift] before every load; serve the
   * tiny pointer table from LDS instead of L2 */
  const bool seg_cached = n_segs <= SEG_LDS;
  if (seg_cached)
    for (int i = threadIdx.x; i < n_segs; i += BS) seg_lds[i] = segs[i];
  const int64_t seg_mask = ((int64_t)1 << seg_shift) - 1;
  const int my = c0 + threadIdx.x;
  uint64_t mykey = GAMMA_KEY_EMPTY;
  if (my < ncand) mykey = keys_in[(size_t)q * ncand + my];
  const bool live = my < ncand && mykey != GAMMA_KEY_EMPTY;
  const uint32_t myid = (uint32_t)(mykey & 0xffffffffu);
  float acc = 0.0f;
  const int f4 = threadIdx.x & 7, rg = threadIdx.x >> 3;
  for (int sl = 0; sl < DV / CH; sl++) {
    __syncthreads();
    for (int r = rg; r < BS; r += BS / 8) {
      int cand = c0 + r;
      if (cand < ncand) {
        uint64_t key = keys_in[(size_t)q * ncand + cand];
        if (key != GAMMA_KEY_EMPTY) {
          uint32_t id = (uint32_t)(key & 0xffffffffu);
          const float *v = (seg_cached ? seg_lds[id >> seg_shift]
                                       : segs[id >> seg_shift]) +
                           (size_t)(id & seg_mask) * DV;
          float4 x = *(const float4 *)(v + sl * CH + f4 * 4);
          rows[r][f4 * 4 + 0] = x.x;
          rows[r][f4 * 4 + 1] = x.y;
          rows[r][f4 * 4 + 2] = x.z;
          rows[r][f4 * 4 + 3] = x.w;
        }
      }
    }
    __syncthreads();
    if (live) {
      const float *qm = qs + sl * CH;
      const float *vm = rows[threadIdx.x];
#pragma unroll
      for (int t = 0; t < CH; t++) {
        if (IP) {
          acc = fmaf(qm[t], vm[t], acc);
        } else {
          float dx = qm[t] - vm[t];
          acc = fmaf(dx, dx, acc);
        }
      }
    }
  }
  if (my < ncand)
    keys_out[(size_t)q * ncand + my] =
        live ? gamma_make_key<IP>(acc, myid) : GAMMA_KEY_EMPTY;
}

hipError_t gk::rerank(hipStream_t s, int nq, int ncand, int d,
                      const float *queries, const float *const *segs,
                      int n_segs, int seg_shift, bool ip,
                      const uint64_t *keys_in, uint64_t *keys_out) {
  int64_t total = (int64_t)nq * ncand;
  if (total == 0) return hipSuccess;
  if (d == 128 || d == 768) {
    const int BS = 256;
    int nchunk = (ncand + BS - 1) / BS;
    dim3 g((uint32_t)((int64_t)nq * nchunk));
#define GAMMA_LAUNCH_RERANK_LDS(IPV, DVV)                                 \
  k_rerank_lds<IPV, DVV, BS><<<g, dim3(BS), 0, s>>>(                      \
      nq, ncand, queries, segs, n_segs, seg_shift, keys_in, keys_out)
    if (ip) {
      if (d == 128) GAMMA_LAUNCH_RERANK_LDS(true, 128);
      else GAMMA_LAUNCH_RERANK_LDS(true, 768);
    } else {
      if (d == 128) GAMMA_LAUNCH_RERANK_LDS(false, 128);
      else GAMMA_LAUNCH_RERANK_LDS(false, 768);
    }
#undef GAMMA_LAUNCH_RERANK_LDS
    return hipGetLastError();
  }
  int64_t blocks = (total + WG - 1) / WG;
#define GAMMA_LAUNCH_RERANK(IPV, DVV)                                     \
  k_rerank<IPV, DVV><<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(         \
      nq, ncand, d, queries, segs, seg_shift, keys_in, keys_out)
  if (ip) {
    GAMMA_LAUNCH_RERANK(true, 0);
  } else {
    GAMMA_LAUNCH_RERANK(false, 0);
  }
#undef GAMMA_LAUNCH_RERANK
  return hipGetLastError();
}

/* ------------------------------------------------------------ row sort */
template <bool IP>
__global__ void k_sort_rows(int nq, int ncand, int k,
                            const uint64_t *__restrict__ keys,
                            float *__restrict__ out_dists,
                            int64_t *__restrict__ out_ids) {
  extern __shared__ char smem[];
  uint64_t *buf = (uint64_t *)smem;
  const int q = blockIdx.x;
  if (q >= nq) return;
  int n = 1;
  while (n < ncand || n < k) n <<= 1; /* k <= n so the output reads are
                                       * always inside the padded region */
  for (int i = threadIdx.x; i < n; i += blockDim.x)
    buf[i] = (i < ncand) ? keys[(int64_t)q * ncand + i] : GAMMA_KEY_EMPTY;
  gamma_bitonic_sort(buf, n);
  for (int i = threadIdx.x; i < k; i += blockDim.x) {
    uint64_t key = buf[i];
    out_dists[(int64_t)q * k + i] =
        (key == GAMMA_KEY_EMPTY) ? -1.0f : gamma_key_dist<IP>(key);
    out_ids[(int64_t)q * k + i] = gamma_key_id(key);
  }
}

hipError_t gk::sort_rows(hipStream_t s, int nq, int ncand, int k,
                         const uint64_t *keys, bool ip, float *out_dists,
                         int64_t *out_ids) {
  int n = 1;
  while (n < ncand || n < k) n <<= 1;
  size_t smem = (size_t)n * 8;
  if (smem > 160 * 1024) return hipErrorInvalidValue;
  if (ip)
    k_sort_rows<true><<<dim3(nq), dim3(WG), smem, s>>>(nq, ncand, k, keys,
                                                       out_dists, out_ids);
  else
    k_sort_rows<false><<<dim3(nq), dim3(WG), smem, s>>>(nq, ncand, k, keys,
                                                        out_dists, out_ids);
  return hipGetLastError();
}

/* ----------------------------------------------------------- unpack keys */
template <bool IP>
__global__ void k_unpack(int64_t n, const uint64_t *__restrict__ keys,
                         float *__restrict__ out_dists,
                         int64_t *__restrict__ out_ids) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t key = keys[i];
  out_dists[i] = (key == GAMMA_KEY_EMPTY) ? -1.0f : gamma_key_dist<IP>(key);
  out_ids[i] = gamma_key_id(key);
}

hipError_t gk::unpack_keys(hipStream_t s, int64_t n, const uint64_t *keys,
                           bool ip, float *out_dists, int64_t *out_ids) {
  if (n == 0) return hipSuccess;
  int64_t blocks = (n + WG - 1) / WG;
  if (ip)
    k_unpack<true><<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(
        n, keys, out_dists, out_ids);
  else
    k_unpack<false><<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(
        n, keys, out_dists, out_ids);
  return hipGetLastError();
}

/* ---------------------------------------------------- pct1 ADC tables */
__global__ void k_pq_btable(int d, int M, int nlist,
                            const float *__restrict__ centroids,
                            const float *__restrict__ codebooks,
                            float *__restrict__ btab) {
  int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)nlist * M * 256;
  if (e >= total) return;
  int dsub = d / M;
  int j = (int)(e & 255);
  int m = (int)((e >> 8) % M);
  int64_t ln = e / ((int64_t)M * 256);
  const float *cm = centroids + ln * d + m * dsub;
  const float *cw = codebooks + ((size_t)m * 256 + j) * dsub;
  float dot = 0.0f;
  if ((dsub & 3) == 0) {
#pragma unroll 2
    for (int t = 0; t < dsub; t += 4) {
      float4 c4 = *(const float4 *)(cw + t);
      float4 m4 = *(const float4 *)(cm + t);
      dot = fmaf(m4.x, c4.x, dot);
      dot = fmaf(m4.y, c4.y, dot);
      dot = fmaf(m4.z, c4.z, dot);
      dot = fmaf(m4.w, c4.w, dot);
    }
  } else {
    for (int t = 0; t < dsub; t++) dot = fmaf(cm[t], cw[t], dot);
  }
  btab[e] = 2.0f * dot;
}

hipError_t gk::pq_tables_b(hipStream_t s, int d, int M, int nlist,
                           const float *centroids, const float *codebooks,
                           float *btab) {
  int64_t total = (int64_t)nlist * M * 256;
  k_pq_btable<<<dim3((uint32_t)((total + WG - 1) / WG)), dim3(WG), 0, s>>>(
      d, M, nlist, centroids, codebooks, btab);
  return hipGetLastError();
}

__global__ void k_pq_atable(int nq, int d, int M,
                            const float *__restrict__ queries,
                            const float *__restrict__ codebooks,
                            float *__restrict__ atab) {
  int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)nq * M * 256;
  if (e >= total) return;
  int dsub = d / M;
  int j = (int)(e & 255);
  int m = (int)((e >> 8) % M);
  int64_t q = e / ((int64_t)M * 256);
  const float *qm = queries + q * d + m * dsub;
  const float *cw = codebooks + ((size_t)m * 256 + j) * dsub;
  float cwn = 0.0f, dot = 0.0f;
  if ((dsub & 3) == 0) { /* float4 loads, same fmaf order */
#pragma unroll 2
    for (int t = 0; t < dsub; t += 4) {
      float4 c4 = *(const float4 *)(cw + t);
      float4 q4 = *(const float4 *)(qm + t);
      cwn = fmaf(c4.x, c4.x, cwn);
      cwn = fmaf(c4.y, c4.y, cwn);
      cwn = fmaf(c4.z, c4.z, cwn);
      cwn = fmaf(c4.w, c4.w, cwn);
      dot = fmaf(q4.x, c4.x, dot);
      dot = fmaf(q4.y, c4.y, dot);
      dot = fmaf(q4.z, c4.z, dot);
      dot = fmaf(q4.w, c4.w, dot);
    }
  } else {
    for (int t = 0; t < dsub; t++) cwn = fmaf(cw[t], cw[t], cwn);
    for (int t = 0; t < dsub; t++) dot = fmaf(qm[t], cw[t], dot);
  }
  atab[e] = fmaf(-2.0f, dot, cwn);
}

hipError_t gk::pq_tables_a(hipStream_t s, int nq, int d, int M,
                           const float *queries, const float *codebooks,
                           float *atab) {
  int64_t total = (int64_t)nq * M * 256;
  k_pq_atable<<<dim3((uint32_t)((total + WG - 1) / WG)), dim3(WG), 0, s>>>(
      nq, d, M, queries, codebooks, atab);
  return hipGetLastError();
}

/* per-vector S term: out[i] = sum_m btab[asg_i][m][code_i[m]], plain
 * adds in m order (the oracle mirrors this exactly —
 * oracle_ivfpq_search_pct1's per-code B sum) */
__global__ void k_pq_sterm(int64_t n, int M, int nlist,
                           const uint8_t *__restrict__ codes,
                           const int32_t *__restrict__ asg, int asg_const,
                           const float *__restrict__ btab,
                           float *__restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int ln = asg ? asg[i] : asg_const;
  if (ln < 0 || ln >= nlist) { /* defensive: matches the host-side
                                  bucket fixup for degenerate
                                  assignments (core.cpp add) */
    out[i] = 0.0f;
    return;
  }
  const float *B = btab + (size_t)ln * M * 256;
  const uint8_t *c = codes + (size_t)i * M;
  float acc = 0.0f;
  for (int m = 0; m < M; m++) acc += B[(size_t)m * 256 + c[m]];
  out[i] = acc;
}

/* Load-path batch: rebuild every bucket's S terms in one launch
 * (block per bucket) instead of one k_pq_sterm launch per bucket —
 * 16384 launches / 0.27 s of Load time at the headline nlist. */
__global__ void k_pq_sterm_buckets(int nlist, int M,
                                   const GammaBucketDev *__restrict__ bks,
                                   const float *__restrict__ btab) {
  const GammaBucketDev bk = bks[blockIdx.x];
  if (!bk.svals || bk.size <= 0) return;
  const float *B = btab + (size_t)blockIdx.x * M * 256;
  const uint8_t *codes = (const uint8_t *)bk.data;
  float *sv = (float *)bk.svals; /* writable: Load owns the buckets */
  for (long long j = threadIdx.x; j < bk.size; j += blockDim.x) {
    const uint8_t *c = codes + (size_t)j * M;
    float acc = 0.0f;
    for (int m = 0; m < M; m++) acc += B[(size_t)m * 256 + c[m]];
    sv[j] = acc;
  }
}

hipError_t gk::pq_sterm_buckets(hipStream_t s, int nlist, int M,
                                const GammaBucketDev *bks,
                                const float *btab) {
  if (nlist <= 0) return hipSuccess;
  k_pq_sterm_buckets<<<dim3((uint32_t)nlist), dim3(256), 0, s>>>(
      nlist, M, bks, btab);
  return hipGetLastError();
}

hipError_t gk::pq_sterm(hipStream_t s, int64_t n, int M, int nlist,
                        const uint8_t *codes, const int32_t *asg,
                        int asg_const, const float *btab, float *out) {
  if (n <= 0) return hipSuccess;
  k_pq_sterm<<<dim3((uint32_t)((n + WG - 1) / WG)), dim3(WG), 0, s>>>(
      n, M, nlist, codes, asg, asg_const, btab, out);
  return hipGetLastError();
}

/* ------------------------------------------------------ bucket scatter */
__global__ void k_bucket_scatter(int nseg,
                                 const GammaScatterSeg *__restrict__ segs,
                                 const uint32_t *__restrict__ ids_src,
                                 const uint8_t *__restrict__ data_src,
                                 const float *__restrict__ svals_src,
                                 int entry_bytes) {
  const GammaScatterSeg sg = segs[blockIdx.x];
  for (long long i = threadIdx.x; i < sg.count; i += blockDim.x) {
    sg.ids_dst[i] = ids_src[sg.src_start + i];
    if (sg.sval_dst && svals_src)
      sg.sval_dst[i] = svals_src[sg.src_start + i];
  }
  /* entry payloads as u32 words when aligned, bytes otherwise */
  long long total = sg.count * entry_bytes;
  const uint8_t *src = data_src + sg.src_start * entry_bytes;
  if ((entry_bytes & 3) == 0) {
    const uint32_t *s4 = (const uint32_t *)src;
    uint32_t *d4 = (uint32_t *)sg.data_dst;
    for (long long i = threadIdx.x; i < (total >> 2); i += blockDim.x)
      d4[i] = s4[i];
  } else {
    for (long long i = threadIdx.x; i < total; i += blockDim.x)
      sg.data_dst[i] = src[i];
  }
}

hipError_t gk::bucket_scatter(hipStream_t s, int nseg,
                              const GammaScatterSeg *segs_dev,
                              const uint32_t *ids_src,
                              const uint8_t *data_src,
                              const float *svals_src, int entry_bytes) {
  if (nseg <= 0) return hipSuccess;
  k_bucket_scatter<<<dim3((uint32_t)nseg), dim3(256), 0, s>>>(
      nseg, segs_dev, ids_src, data_src, svals_src, entry_bytes);
  return hipGetLastError();
}

/* ----------------------------------------------------------- residuals */
__global__ void k_residuals(int64_t n, int d, const float *__restrict__ x,
                            const float *__restrict__ centroids,
                            const int32_t *__restrict__ assign,
                            float *__restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n * d) return;
  int64_t row = i / d;
  int col = (int)(i - row * d);
  out[i] = x[i] - centroids[(size_t)assign[row] * d + col];
}

hipError_t gk::residuals(hipStream_t s, int64_t n, int d, const float *x,
                         const float *centroids, const int32_t *assign,
                         float *out) {
  if (n == 0) return hipSuccess;
  int64_t blocks = (n * d + WG - 1) / WG;
  k_residuals<<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(n, d, x, centroids,
                                                          assign, out);
  return hipGetLastError();
}

/* ----------------------------------------------------------- PQ encode */
__global__ void k_pq_encode(int64_t n, int d, int M, int ksub,
                            const float *__restrict__ x,
                            const float *__restrict__ codebooks,
                            uint8_t *__restrict__ codes) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n * M) return;
  int64_t i = idx / M;
  int m = (int)(idx - i * M);
  int dsub = d / M;
  const float *xm = x + i * d + m * dsub;
  const float *cb = codebooks + (size_t)m * ksub * dsub;
  float best = INFINITY;
  int bestj = 0;
  for (int j = 0; j < ksub; j++) {
    const float *cw = cb + (size_t)j * dsub;
    float acc = 0.0f;
    for (int t = 0; t < dsub; t++) {
      float diff = xm[t] - cw[t];
      acc = fmaf(diff, diff, acc);
    }
    if (acc < best) { best = acc; bestj = j; }
  }
  codes[idx] = (uint8_t)bestj;
}

hipError_t gk::pq_encode(hipStream_t s, int64_t n, int d, int M, int ksub,
                         const float *x, const float *codebooks,
                         uint8_t *codes) {
  if (n == 0) return hipSuccess;
  int64_t blocks = (n * M + WG - 1) / WG;
  k_pq_encode<<<dim3((uint32_t)blocks), dim3(WG), 0, s>>>(n, d, M, ksub, x,
                                                          codebooks, codes);
  return hipGetLastError();
}
