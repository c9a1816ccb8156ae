// NOTE: r1-era harness — replicates the ROUND-1 scan formulation
// (per-list T = A + B staging, pre S-term). Kept so the r1 ablation
// record (profiles/r01_adc_ablation_selectors.txt) stays reproducible;
// the production kernel has since moved to the per-vector S term.
/*
 * adc_bench.hip — standalone ablation bench for the IVFPQ ADC scan kernel
 * (north-star shape: d=128, M=32, nlist=4096, N=10M, nprobe=32, nq
 * configurable). Builds synthetic buckets directly on the GPU and times
 * kernel variants; used to find where the cycles go (§5 common-mistake 8:
 * ablate before optimizing). Not part of the product path.
 *
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/adc_bench.hip
 *        -I vearch_amd/csrc -o /tmp/adc_bench
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>

#include "../vearch_amd/csrc/kernels.h"
#include "../vearch_amd/csrc/select.hpp"

#define CHECK(x)                                                      \
  do {                                                                \
    hipError_t e = (x);                                               \
    if (e != hipSuccess) {                                            \
      printf("HIP error %s at line %d\n", hipGetErrorString(e),       \
             __LINE__);                                               \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

/* variant bits */
#define V_NOPUSH 1   /* skip selector (keep dis live via asm) */
#define V_NOLUT 2    /* skip per-list LUT build (use list 0's) */
#define V_NOSCAN 4   /* skip the code scan (LUT build only) */
#define V_PCT1 8     /* decomposed tables: T = A_q + B_list (ivfpq.h:254) */
#define V_BITMAP 16  /* test the delete bitmap per code */
#define V_PCT1V 32   /* pct1 with float4-vectorized table build */
#define V_FASTCMP 64 /* float-key compare before the full push */
#define V_U32ID 128  /* 4-byte ids (bit 31 delete) instead of int64 */
#define V_WSEL 256   /* wave-local selector (no cross-wave barriers) */

template <int MW, int C, int VAR, int BS = 256>
__global__ void __launch_bounds__(BS)
k_scan_var(int nq, int d, int M, int nprobe, int k2,
           const float *__restrict__ queries,
           const float *__restrict__ centroids,
           const float *__restrict__ codebooks,
           const GammaBucketDev *__restrict__ buckets, int nlist,
           const int64_t *__restrict__ probes,
           uint64_t *__restrict__ out_keys,
           const float *__restrict__ Atab,   /* nq x M*ksub */
           const float *__restrict__ Btab,   /* nlist x M*ksub */
           const uint32_t *__restrict__ bitmap) {
  extern __shared__ char smem[];
  const int ksub = 256;
  const int dsub = d / M;
  float *lut = (float *)smem;
  uint64_t *sortbuf = (uint64_t *)(smem + ((size_t)M * ksub * 4 + 7) / 8 * 8);
  uint64_t *res = sortbuf + GAMMA_SORT_CAP;
  float *qs = (float *)(res + k2);
  float *rs = qs + d;
  int *state = (int *)(rs + d);

  const int q = blockIdx.x;
  if (q >= nq) return;
  const float *qg = queries + (int64_t)q * d;
  for (int i = threadIdx.x; i < d; i += blockDim.x) qs[i] = qg[i];

  GammaSelector sel;
  sel.init(sortbuf, res, state, k2);

  for (int p = 0; p < nprobe; p++) {
    int64_t ln = probes[(int64_t)q * nprobe + p];
    if (ln < 0 || ln >= nlist) continue;
    GammaBucketDev bk = buckets[ln];
    if (bk.size <= 0) continue;
    const float *cent = centroids + (size_t)ln * d;

    if (VAR & V_PCT1V) {
      const float4 *Aq =
          (const float4 *)(Atab + (size_t)q * M * ksub);
      const float4 *Bl =
          (const float4 *)(Btab + (size_t)ln * M * ksub);
      float4 *lut4 = (float4 *)lut;
      for (int e = threadIdx.x; e < (M * ksub) / 4; e += blockDim.x) {
        float4 a = Aq[e], b = Bl[e];
        lut4[e] = make_float4(a.x + b.x, a.y + b.y, a.z + b.z, a.w + b.w);
      }
      __syncthreads();
    } else if (VAR & V_PCT1) {
      const float *Aq = Atab + (size_t)q * M * ksub;
      const float *Bl = Btab + (size_t)ln * M * ksub;
      for (int e = threadIdx.x; e < M * ksub; e += blockDim.x)
        lut[e] = Aq[e] + Bl[e];
      __syncthreads();
    } else if (!(VAR & V_NOLUT) || p == 0) {
      for (int i = threadIdx.x; i < d; i += blockDim.x)
        rs[i] = qs[i] - cent[i];
      __syncthreads();
      for (int e = threadIdx.x; e < M * ksub; e += blockDim.x) {
        int m = e >> 8, j = e & 255;
        const float *cw = codebooks + ((size_t)m * ksub + j) * dsub;
        const float *rm = rs + m * dsub;
        float acc = 0.0f;
        for (int t = 0; t < dsub; t++) {
          float diff = rm[t] - cw[t];
          acc = fmaf(diff, diff, acc);
        }
        lut[e] = acc;
      }
      __syncthreads();
    }
    if (VAR & V_NOSCAN) continue;

    const uint32_t *ids32 = bk.ids;
    const uint8_t *codes = (const uint8_t *)bk.data;
    for (long long j0 = 0; j0 < bk.size; j0 += (long long)blockDim.x * C) {
      long long jb = j0 + (long long)threadIdx.x * C;
      uint32_t w[C][MW];
      int64_t idv[C];
#pragma unroll
      for (int c = 0; c < C; c++) {
        long long j = jb + c;
        if (j < bk.size) {
          idv[c] = (int64_t)(int32_t)ids32[j];
          const uint32_t *cw = (const uint32_t *)(codes + (size_t)j * M);
#pragma unroll
          for (int mw = 0; mw < MW; mw++) w[c][mw] = cw[mw];
        } else {
          idv[c] = -1;
        }
      }
#pragma unroll
      for (int c = 0; c < C; c++) {
        int64_t id = idv[c];
        if (!((uint64_t)id >> 63) &&
            (!(VAR & V_BITMAP) ||
             !gamma_bitmap_test(bitmap, (uint64_t)id))) {
          float dis = 0.0f;
          const float *tab = lut;
#pragma unroll
          for (int mw = 0; mw < MW; mw++) {
            uint32_t wv = w[c][mw];
            dis += tab[wv & 255u];         tab += ksub;
            dis += tab[(wv >> 8) & 255u];  tab += ksub;
            dis += tab[(wv >> 16) & 255u]; tab += ksub;
            dis += tab[wv >> 24];          tab += ksub;
          }
          if (VAR & V_NOPUSH) {
            asm volatile("" ::"v"(dis)); /* keep the work live */
          } else if (VAR & V_FASTCMP) {
            uint32_t dk = gamma_f32_key(dis);
            if (dk < (uint32_t)(sel.thresh >> 32) ||
                (dk == (uint32_t)(sel.thresh >> 32) &&
                 (uint32_t)id < (uint32_t)sel.thresh)) {
              int idx = atomicAdd(&sel.state[0], 1);
              sel.sortbuf[idx] = ((uint64_t)dk << 32) | (uint32_t)id;
            }
          } else {
            sel.push(gamma_make_key<false>(dis, (uint32_t)id));
          }
        }
      }
      if (!(VAR & V_NOPUSH)) sel.maybe_flush(blockDim.x * C);
    }
    __syncthreads();
  }
  sel.finish();
  for (int i = threadIdx.x; i < k2; i += blockDim.x)
    out_keys[(int64_t)q * k2 + i] = res[i];
}

__global__ void k_fill(uint8_t *codes, uint32_t *ids, int64_t n, int M,
                       uint64_t seed) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  ids[i] = (uint32_t)i;
  uint64_t s = seed + i * 2654435761ull;
  for (int m = 0; m < M; m++) {
    s = s * 6364136223846793005ull + 1442695040888963407ull;
    codes[i * M + m] = (uint8_t)(s >> 33);
  }
}

__global__ void k_fillf(float *p, int64_t n, uint64_t seed) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t s = (seed + i) * 2654435761ull;
  s = s * 6364136223846793005ull + 1442695040888963407ull;
  p[i] = (float)((s >> 40) & 0xFFFF) / 65536.0f - 0.5f;
}

/* Wave-selector scan: pct1v LUT + u32 ids + GammaWaveSelector. Each
 * wave keeps its own top-k2 in a private LDS region; the only
 * block-wide barriers are around the per-list LUT build. Final merge:
 * compact nw*k2 partials, pad, one block bitonic. */
template <int MW, int C, int BS>
__global__ void __launch_bounds__(BS)
k_scan_wsel(int nq, int d, int M, int nprobe, int k2,
            const float *__restrict__ queries,
            const float *__restrict__ centroids,
            const float *__restrict__ codebooks,
            const GammaBucketDev *__restrict__ buckets, int nlist,
            const int64_t *__restrict__ probes,
            uint64_t *__restrict__ out_keys,
            const float *__restrict__ Atab,
            const float *__restrict__ Btab,
            const uint32_t *__restrict__ bitmap) {
  extern __shared__ char smem[];
  const int ksub = 256;
  const int nw = BS / 64;
  float *lut = (float *)smem;
  uint64_t *wbase = (uint64_t *)(smem + ((size_t)M * ksub * 4 + 7) / 8 * 8);
  int *cnts = (int *)(wbase + (size_t)nw * GAMMA_WSEL_CAP);

  const int q = blockIdx.x;
  if (q >= nq) return;
  const int wave = threadIdx.x >> 6;

  GammaWaveSelector wsel;
  wsel.init(wbase + (size_t)wave * GAMMA_WSEL_CAP, cnts + wave, k2);

  for (int p = 0; p < nprobe; p++) {
    int64_t ln = probes[(int64_t)q * nprobe + p];
    if (ln < 0 || ln >= nlist) continue;
    GammaBucketDev bk = buckets[ln];
    if (bk.size <= 0) continue;

    __syncthreads(); /* previous list's LUT reads done before overwrite */
    const float4 *Aq = (const float4 *)(Atab + (size_t)q * M * ksub);
    const float4 *Bl = (const float4 *)(Btab + (size_t)ln * M * ksub);
    float4 *lut4 = (float4 *)lut;
    for (int e = threadIdx.x; e < (M * ksub) / 4; e += blockDim.x) {
      float4 a = Aq[e], b = Bl[e];
      lut4[e] = make_float4(a.x + b.x, a.y + b.y, a.z + b.z, a.w + b.w);
    }
    __syncthreads();

    const uint32_t *ids32 = bk.ids;
    const uint8_t *codes = (const uint8_t *)bk.data;
    for (long long j0 = 0; j0 < bk.size; j0 += (long long)blockDim.x * C) {
      long long jb = j0 + (long long)threadIdx.x * C;
      uint32_t w[C][MW];
      int64_t idv[C];
#pragma unroll
      for (int c = 0; c < C; c++) {
        long long j = jb + c;
        if (j < bk.size) {
          idv[c] = (int64_t)(int32_t)ids32[j];
          const uint32_t *cw = (const uint32_t *)(codes + (size_t)j * M);
#pragma unroll
          for (int mw = 0; mw < MW; mw++) w[c][mw] = cw[mw];
        } else {
          idv[c] = -1;
        }
      }
#pragma unroll
      for (int c = 0; c < C; c++) {
        int64_t id = idv[c];
        if (!((uint64_t)id >> 63)) {
          float dis = 0.0f;
          const float *tab = lut;
#pragma unroll
          for (int mw = 0; mw < MW; mw++) {
            uint32_t wv = w[c][mw];
            dis += tab[wv & 255u];         tab += ksub;
            dis += tab[(wv >> 8) & 255u];  tab += ksub;
            dis += tab[(wv >> 16) & 255u]; tab += ksub;
            dis += tab[wv >> 24];          tab += ksub;
          }
          wsel.push(gamma_make_key<false>(dis, (uint32_t)id));
        }
      }
      wsel.maybe_flush(64 * C); /* per-wave; no block barrier */
    }
  }
  wsel.finish();
  __syncthreads();

  /* compact the nw sorted partials [w*CAP, w*CAP+k2) into [0, nw*k2):
   * read-all-then-write (regions overlap), <=4 elems per thread */
  const int tot = nw * k2;
  const int tid = threadIdx.x;
  uint64_t v0 = GAMMA_KEY_EMPTY, v1 = GAMMA_KEY_EMPTY,
           v2 = GAMMA_KEY_EMPTY, v3 = GAMMA_KEY_EMPTY;
  {
    int e0 = tid, e1 = tid + BS, e2 = tid + 2 * BS, e3 = tid + 3 * BS;
    if (e0 < tot) v0 = wbase[(e0 / k2) * GAMMA_WSEL_CAP + e0 % k2];
    if (e1 < tot) v1 = wbase[(e1 / k2) * GAMMA_WSEL_CAP + e1 % k2];
    if (e2 < tot) v2 = wbase[(e2 / k2) * GAMMA_WSEL_CAP + e2 % k2];
    if (e3 < tot) v3 = wbase[(e3 / k2) * GAMMA_WSEL_CAP + e3 % k2];
    __syncthreads();
    if (e0 < tot) wbase[e0] = v0;
    if (e1 < tot) wbase[e1] = v1;
    if (e2 < tot) wbase[e2] = v2;
    if (e3 < tot) wbase[e3] = v3;
  }
  int n2 = 1;
  while (n2 < tot) n2 <<= 1;
  __syncthreads();
  for (int i = tot + tid; i < n2; i += BS) wbase[i] = GAMMA_KEY_EMPTY;
  gamma_bitonic_sort(wbase, n2);
  for (int i = tid; i < k2; i += BS)
    out_keys[(int64_t)q * k2 + i] = wbase[i];
}

int main(int argc, char **argv) {
  int nq = argc > 1 ? atoi(argv[1]) : 10000;
  const int nprobe = argc > 2 ? atoi(argv[2]) : 32;
  const int d = 128, M = 32, MW = 8, nlist = 4096, k2 = 200;
  const int64_t N = 10000000;
  const int64_t per = N / nlist;

  float *queries, *centroids, *codebooks;
  CHECK(hipMalloc(&queries, (size_t)nq * d * 4));
  CHECK(hipMalloc(&centroids, (size_t)nlist * d * 4));
  CHECK(hipMalloc(&codebooks, (size_t)M * 256 * (d / M) * 4));
  /* one big slab for codes+ids, sliced into buckets */
  uint8_t *codes;
  uint32_t *ids;
  CHECK(hipMalloc(&codes, (size_t)N * M));
  CHECK(hipMalloc(&ids, (size_t)N * 4));
  k_fill<<<dim3((uint32_t)((N + 255) / 256)), dim3(256)>>>(codes, ids, N, M,
                                                           42);
  std::vector<GammaBucketDev> hb(nlist);
  for (int i = 0; i < nlist; i++) {
    hb[i].ids = ids + (size_t)i * per;
    hb[i].data = codes + (size_t)i * per * M;
    hb[i].size = per;
  }
  GammaBucketDev *buckets;
  CHECK(hipMalloc(&buckets, nlist * sizeof(GammaBucketDev)));
  CHECK(hipMemcpy(buckets, hb.data(), nlist * sizeof(GammaBucketDev),
                  hipMemcpyHostToDevice));
  int64_t *probes;
  CHECK(hipMalloc(&probes, (size_t)nq * nprobe * 8));
  std::vector<int64_t> hp((size_t)nq * nprobe);
  srand(7);
  for (size_t i = 0; i < hp.size(); i++) hp[i] = rand() % nlist;
  CHECK(hipMemcpy(probes, hp.data(), hp.size() * 8,
                  hipMemcpyHostToDevice));
  uint64_t *out;
  CHECK(hipMalloc(&out, (size_t)nq * k2 * 8));
  float *Atab, *Btab;
  CHECK(hipMalloc(&Atab, (size_t)nq * M * 256 * 4));
  CHECK(hipMalloc(&Btab, (size_t)nlist * M * 256 * 4));
  k_fillf<<<dim3((uint32_t)(((size_t)nq * M * 256 + 255) / 256)),
            dim3(256)>>>(Atab, (int64_t)nq * M * 256, 11);
  k_fillf<<<dim3((uint32_t)(((size_t)nlist * M * 256 + 255) / 256)),
            dim3(256)>>>(Btab, (int64_t)nlist * M * 256, 13);
  uint32_t *bitmap;
  CHECK(hipMalloc(&bitmap, (size_t)(N + 31) / 32 * 4));
  CHECK(hipMemset(bitmap, 0, (size_t)(N + 31) / 32 * 4));
  CHECK(hipMemset(queries, 1, (size_t)nq * d * 4));
  CHECK(hipMemset(centroids, 2, (size_t)nlist * d * 4));
  CHECK(hipMemset(codebooks, 3, (size_t)M * 256 * (d / M) * 4));

  size_t smem = ((size_t)M * 256 * 4 + 7) / 8 * 8 +
                (GAMMA_SORT_CAP + k2) * 8 + (2 * d + 1) * 4 + 16;
  double bytes_per_q = (double)nprobe * per * (M + 8);

  auto run_bs = [&](const char *name, auto kern, int reps, int bs) {
    /* warmup */
    kern<<<dim3(nq), dim3(bs), smem>>>(nq, d, M, nprobe, k2, queries,
                                       centroids, codebooks, buckets,
                                       nlist, probes, out, Atab, Btab,
                                       bitmap);
    CHECK(hipGetLastError());
    CHECK(hipDeviceSynchronize());
    hipEvent_t a, b;
    hipEventCreate(&a);
    hipEventCreate(&b);
    hipEventRecord(a);
    for (int r = 0; r < reps; r++)
      kern<<<dim3(nq), dim3(bs), smem>>>(nq, d, M, nprobe, k2, queries,
                                         centroids, codebooks, buckets,
                                         nlist, probes, out, Atab, Btab,
                                         bitmap);
    hipEventRecord(b);
    CHECK(hipEventSynchronize(b));
    float ms;
    hipEventElapsedTime(&ms, a, b);
    ms /= reps;
    printf("%-28s %8.2f ms  %8.1f GB/s  %9.0f QPS\n", name, ms,
           nq * bytes_per_q / ms / 1e6, nq / ms * 1000.0);
    hipEventDestroy(a);
    hipEventDestroy(b);
  };
  auto run = [&](const char *name, auto kern, int reps) {
    run_bs(name, kern, reps, 256);
  };
  /* result validation: compare out vs the reference variant's output */
  std::vector<uint64_t> h_ref((size_t)nq * k2), h_got((size_t)nq * k2);
  auto snap = [&](std::vector<uint64_t> &dst) {
    CHECK(hipMemcpy(dst.data(), out, dst.size() * 8,
                    hipMemcpyDeviceToHost));
  };
  auto check = [&](const char *name) {
    snap(h_got);
    size_t bad = 0;
    for (size_t i = 0; i < h_got.size(); i++)
      if (h_got[i] != h_ref[i]) bad++;
    if (bad)
      printf("  MISMATCH %-22s %zu/%zu keys differ\n", name, bad,
             h_got.size());
    else
      printf("  ok       %-22s output == reference\n", name);
  };

  printf("nq=%d N=%lld nlist=%d nprobe=%d M=%d k2=%d smem=%zu\n", nq,
         (long long)N, nlist, nprobe, M, k2, smem);
  run("full C=4", k_scan_var<MW, 4, 0>, 3);
  run("full C=8", k_scan_var<MW, 8, 0>, 3);
  run("full C=2", k_scan_var<MW, 2, 0>, 3);
  run("nopush C=4", k_scan_var<MW, 4, V_NOPUSH>, 3);
  run("nolut C=4", k_scan_var<MW, 4, V_NOLUT>, 3);
  run("lutonly", k_scan_var<MW, 4, V_NOSCAN | V_NOPUSH>, 3);
  run("nopush+nolut C=4", k_scan_var<MW, 4, V_NOPUSH | V_NOLUT>, 3);
  run("nopush+nolut C=8", k_scan_var<MW, 8, V_NOPUSH | V_NOLUT>, 3);
  run("pct1 C=4", k_scan_var<MW, 4, V_PCT1>, 3);
  run("pct1 C=2", k_scan_var<MW, 2, V_PCT1>, 3);
  run("pct1 C=8", k_scan_var<MW, 8, V_PCT1>, 3);
  run("pct1+bitmap C=4", k_scan_var<MW, 4, V_PCT1 | V_BITMAP>, 3);
  run("pct1+nopush C=4", k_scan_var<MW, 4, V_PCT1 | V_NOPUSH>, 3);
  run("pct1v C=4", k_scan_var<MW, 4, V_PCT1V>, 3);
  run("pct1v C=2", k_scan_var<MW, 2, V_PCT1V>, 3);
  run("pct1v+nopush C=2", k_scan_var<MW, 2, V_PCT1V | V_NOPUSH>, 3);
  run("pct1v+fast C=2", k_scan_var<MW, 2, V_PCT1V | V_FASTCMP>, 3);
  run("pct1v+fast C=4", k_scan_var<MW, 4, V_PCT1V | V_FASTCMP>, 3);
  run("pct1v+fast+bm C=2", k_scan_var<MW, 2, V_PCT1V | V_FASTCMP | V_BITMAP>, 3);
  run("pct1v+u32 C=2", k_scan_var<MW, 2, V_PCT1V | V_U32ID>, 3);
  run("pct1v+u32 C=4", k_scan_var<MW, 4, V_PCT1V | V_U32ID>, 3);
  run("pct1v+u32+nopush C=2",
      k_scan_var<MW, 2, V_PCT1V | V_U32ID | V_NOPUSH>, 3);
  run_bs("pct1v+u32 C=2 BS512",
         k_scan_var<MW, 2, V_PCT1V | V_U32ID, 512>, 3, 512);
  snap(h_ref); /* engine config = the validation reference */
  run("pct1v+fast C=2 chk", k_scan_var<MW, 2, V_PCT1V | V_FASTCMP>, 3);
  check("pct1v+fast C=2");
  run_bs("pct1v+u32 C=1 BS512",
         k_scan_var<MW, 1, V_PCT1V | V_U32ID, 512>, 3, 512);
  run_bs("pct1v+u32 C=4 BS512",
         k_scan_var<MW, 4, V_PCT1V | V_U32ID, 512>, 3, 512);
  run_bs("pct1v+u32+nopush BS512",
         k_scan_var<MW, 2, V_PCT1V | V_U32ID | V_NOPUSH, 512>, 3, 512);

  auto run_wsel = [&](const char *name, auto kern, int reps, int bs) {
    int nw = bs / 64;
    size_t wsmem = ((size_t)M * 256 * 4 + 7) / 8 * 8 +
                   (size_t)nw * GAMMA_WSEL_CAP * 8 + nw * 4;
    kern<<<dim3(nq), dim3(bs), wsmem>>>(nq, d, M, nprobe, k2, queries,
                                        centroids, codebooks, buckets,
                                        nlist, probes, out, Atab, Btab,
                                        bitmap);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess) {
      printf("%-28s launch failed: %s (smem=%zu)\n", name,
             hipGetErrorString(e), wsmem);
      return;
    }
    CHECK(hipDeviceSynchronize());
    check(name);
    hipEvent_t a, b;
    (void)hipEventCreate(&a);
    (void)hipEventCreate(&b);
    (void)hipEventRecord(a);
    for (int r = 0; r < reps; r++)
      kern<<<dim3(nq), dim3(bs), wsmem>>>(nq, d, M, nprobe, k2, queries,
                                          centroids, codebooks, buckets,
                                          nlist, probes, out, Atab, Btab,
                                          bitmap);
    (void)hipEventRecord(b);
    CHECK(hipEventSynchronize(b));
    float ms;
    (void)hipEventElapsedTime(&ms, a, b);
    ms /= reps;
    printf("%-28s %8.2f ms  %8.1f GB/s  %9.0f QPS\n", name, ms,
           nq * bytes_per_q / ms / 1e6, nq / ms * 1000.0);
    (void)hipEventDestroy(a);
    (void)hipEventDestroy(b);
  };
  run_wsel("wsel C=2 BS128", k_scan_wsel<MW, 2, 128>, 3, 128);
  run_wsel("wsel C=2 BS256", k_scan_wsel<MW, 2, 256>, 3, 256);
  run_wsel("wsel C=4 BS256", k_scan_wsel<MW, 4, 256>, 3, 256);
  run_wsel("wsel C=1 BS256", k_scan_wsel<MW, 1, 256>, 3, 256);
  return 0;
}
