/*
 * rerank_bench.hip — ablation bench for the exact re-rank kernel
 * (k_rerank): 10k queries x 200 candidates x d=128 fp32 gathers from a
 * 10M-vector store. PMC evidence (profiles/r01_fetch_size_c1.csv)
 * shows the production kernel fetches exactly its algorithmic bytes
 * but only reaches ~1.4 TB/s — latency/occupancy-bound. Variants here
 * probe memory-level-parallelism fixes; every variant's output is
 * compared byte-exactly against V0 (the canonical chain must be
 * preserved — FLAT parity depends on it).
 *
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17
 *        tools/rerank_bench.hip -I vearch_amd/csrc -o /tmp/rerank_bench
 * Not part of the product path.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <vector>

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

/* V0: mirror of the production kernel (k_rerank<false, 128>) */
template <int DV, int UNROLL>
__global__ void k_rr_base(int nq, int ncand, const float *__restrict__ q,
                          const float *__restrict__ store,
                          const uint64_t *__restrict__ keys_in,
                          uint64_t *__restrict__ keys_out) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (int64_t)nq * ncand) return;
  int qq = (int)(idx / ncand);
  uint64_t key = keys_in[idx];
  uint32_t id = (uint32_t)(key & 0xffffffffu);
  const float4 *v4 = (const float4 *)(store + (size_t)id * DV);
  const float4 *q4 = (const float4 *)(q + (size_t)qq * DV);
  float acc = 0.0f;
#pragma unroll UNROLL
  for (int t = 0; t < DV / 4; t++) {
    float4 a = q4[t], b = v4[t];
    float dx = a.x - b.x, dy = a.y - b.y, dz = a.z - b.z, dw = a.w - b.w;
    acc = fmaf(dx, dx, acc);
    acc = fmaf(dy, dy, acc);
    acc = fmaf(dz, dz, acc);
    acc = fmaf(dw, dw, acc);
  }
  keys_out[idx] = gamma_make_key<false>(acc, id);
}

/* V1: two candidates per thread, chains interleaved (each chain stays
 * canonical; 2x loads in flight per thread) */
template <int DV>
__global__ void k_rr_ilp2(int nq, int ncand, const float *__restrict__ q,
                          const float *__restrict__ store,
                          const uint64_t *__restrict__ keys_in,
                          uint64_t *__restrict__ keys_out) {
  int64_t pair = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)nq * ncand;
  int64_t i0 = pair * 2, i1 = pair * 2 + 1;
  if (i0 >= total) return;
  int q0 = (int)(i0 / ncand);
  uint32_t id0 = (uint32_t)(keys_in[i0] & 0xffffffffu);
  const float4 *v0 = (const float4 *)(store + (size_t)id0 * DV);
  const float4 *qa = (const float4 *)(q + (size_t)q0 * DV);
  bool has1 = i1 < total;
  int q1 = has1 ? (int)(i1 / ncand) : q0;
  uint32_t id1 = has1 ? (uint32_t)(keys_in[i1] & 0xffffffffu) : id0;
  const float4 *v1 = (const float4 *)(store + (size_t)id1 * DV);
  const float4 *qb = (const float4 *)(q + (size_t)q1 * DV);
  float acc0 = 0.0f, acc1 = 0.0f;
#pragma unroll 4
  for (int t = 0; t < DV / 4; t++) {
    float4 a0 = qa[t], b0 = v0[t];
    float4 a1 = qb[t], b1 = v1[t];
    float dx = a0.x - b0.x, dy = a0.y - b0.y, dz = a0.z - b0.z,
          dw = a0.w - b0.w;
    acc0 = fmaf(dx, dx, acc0);
    acc0 = fmaf(dy, dy, acc0);
    acc0 = fmaf(dz, dz, acc0);
    acc0 = fmaf(dw, dw, acc0);
    float ex = a1.x - b1.x, ey = a1.y - b1.y, ez = a1.z - b1.z,
          ew = a1.w - b1.w;
    acc1 = fmaf(ex, ex, acc1);
    acc1 = fmaf(ey, ey, acc1);
    acc1 = fmaf(ez, ez, acc1);
    acc1 = fmaf(ew, ew, acc1);
  }
  keys_out[i0] = gamma_make_key<false>(acc0, id0);
  if (has1) keys_out[i1] = gamma_make_key<false>(acc1, id1);
}

/* V2: LDS-staged rows. One block = one query's candidate chunk of
 * BS rows; d processed in 32-float slices staged cooperatively
 * (8 lanes x float4 = 128 B contiguous per row), each thread then
 * accumulates its own canonical chain from LDS. */
template <int DV, int BS>
__global__ void __launch_bounds__(BS)
k_rr_lds(int nq, int ncand, const float *__restrict__ q,
         const float *__restrict__ store,
         const uint64_t *__restrict__ keys_in,
         uint64_t *__restrict__ keys_out) {
  const int CH = 32; /* floats per slice */
  __shared__ float rows[BS][CH + 1]; /* +1: bank-shift */
  __shared__ float qs[DV];
  const int nchunk = (ncand + BS - 1) / BS;
  const int qq = blockIdx.x / nchunk;
  const int c0 = (blockIdx.x % nchunk) * BS;
  if (qq >= nq) return;
  for (int i = threadIdx.x; i < DV; i += BS)
    qs[i] = q[(size_t)qq * DV + i];
  const int my = c0 + threadIdx.x; /* my candidate */
  uint32_t myid = 0;
  bool live = my < ncand;
  if (live) myid = (uint32_t)(keys_in[(size_t)qq * ncand + my] &
                              0xffffffffu);
  float acc = 0.0f;
  for (int s = 0; s < DV / CH; s++) {
    __syncthreads();
    /* stage BS rows x 128 B: thread t loads f4 (t&7) of row c0+(t>>3),
     * iterating over row groups of BS/8 */
    const int f4 = threadIdx.x & 7, rg = threadIdx.x >> 3;
    for (int r = rg; r < BS; r += BS / 8) {
      int cand = c0 + r;
      if (cand < ncand) {
        uint32_t id = (uint32_t)(keys_in[(size_t)qq * ncand + cand] &
                                 0xffffffffu);
        float4 v = *(const float4 *)(store + (size_t)id * DV + s * CH +
                                     f4 * 4);
        rows[r][f4 * 4 + 0] = v.x;
        rows[r][f4 * 4 + 1] = v.y;
        rows[r][f4 * 4 + 2] = v.z;
        rows[r][f4 * 4 + 3] = v.w;
      }
    }
    __syncthreads();
    if (live) {
      const float *qm = qs + s * CH;
      const float *vm = rows[threadIdx.x];
#pragma unroll
      for (int t = 0; t < CH; t++) {
        float dx = qm[t] - vm[t];
        acc = fmaf(dx, dx, acc);
      }
    }
  }
  if (live)
    keys_out[(size_t)qq * ncand + my] = gamma_make_key<false>(acc, myid);
}

__global__ void k_fill(float *p, int64_t n, uint64_t seed) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t s = (seed + i) * 2654435761ull;
  s = s * 6364136223846793005ull + 1442695040888963407ull;
  p[i] = (float)((s >> 40) & 0xFFFF) / 65536.0f - 0.5f;
}

__global__ void k_fill_keys(uint64_t *k, int64_t n, int64_t nstore,
                            uint64_t seed) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t s = (seed + i) * 2654435761ull;
  s = s * 6364136223846793005ull + 1442695040888963407ull;
  k[i] = ((uint64_t)i << 32) | (uint32_t)(s % nstore);
}

int main(int argc, char **argv) {
  const int d = 128, nq = argc > 1 ? atoi(argv[1]) : 10000;
  const int ncand = argc > 2 ? atoi(argv[2]) : 200;
  const int64_t N = 10000000;
  float *store, *q;
  uint64_t *kin, *kout;
  CHECK(hipMalloc(&store, (size_t)N * d * 4));
  CHECK(hipMalloc(&q, (size_t)nq * d * 4));
  CHECK(hipMalloc(&kin, (size_t)nq * ncand * 8));
  CHECK(hipMalloc(&kout, (size_t)nq * ncand * 8));
  k_fill<<<dim3((uint32_t)((N * d + 255) / 256)), dim3(256)>>>(
      store, N * d, 1);
  k_fill<<<dim3((uint32_t)(((int64_t)nq * d + 255) / 256)), dim3(256)>>>(
      q, (int64_t)nq * d, 2);
  k_fill_keys<<<dim3((uint32_t)(((int64_t)nq * ncand + 255) / 256)),
                dim3(256)>>>(kin, (int64_t)nq * ncand, N, 3);
  CHECK(hipDeviceSynchronize());

  int64_t total = (int64_t)nq * ncand;
  std::vector<uint64_t> ref(total), got(total);
  double bytes = (double)total * (d * 4 + 8) + (double)nq * d * 4;

  auto run = [&](const char *name, auto launch, bool is_ref) {
    launch(); /* warm */
    CHECK(hipGetLastError());
    CHECK(hipDeviceSynchronize());
    if (is_ref) {
      CHECK(hipMemcpy(ref.data(), kout, total * 8,
                      hipMemcpyDeviceToHost));
    } else {
      CHECK(hipMemcpy(got.data(), kout, total * 8,
                      hipMemcpyDeviceToHost));
      size_t bad = 0;
      for (int64_t i = 0; i < total; i++)
        if (got[i] != ref[i]) bad++;
      if (bad) {
        printf("  MISMATCH %-20s %zu/%lld keys differ\n", name, bad,
               (long long)total);
        return;
      }
      printf("  ok       %-20s == reference\n", name);
    }
    hipEvent_t a, b;
    (void)hipEventCreate(&a);
    (void)hipEventCreate(&b);
    (void)hipEventRecord(a);
    for (int r = 0; r < 10; r++) launch();
    (void)hipEventRecord(b);
    CHECK(hipEventSynchronize(b));
    float ms;
    (void)hipEventElapsedTime(&ms, a, b);
    ms /= 10;
    printf("%-28s %8.3f ms  %8.1f GB/s algorithmic\n", name, ms,
           bytes / ms / 1e6);
    (void)hipEventDestroy(a);
    (void)hipEventDestroy(b);
  };

  printf("rerank bench: nq=%d ncand=%d d=%d store=%lldx%d\n", nq, ncand,
         d, (long long)N, d);
  int64_t blocks = (total + 255) / 256;
  run("v0 unroll8 (production)", [&] {
    k_rr_base<128, 8><<<dim3((uint32_t)blocks), dim3(256)>>>(
        nq, ncand, q, store, kin, kout);
  }, true);
  run("v0 unroll4", [&] {
    k_rr_base<128, 4><<<dim3((uint32_t)blocks), dim3(256)>>>(
        nq, ncand, q, store, kin, kout);
  }, false);
  run("v0 unroll32 (full)", [&] {
    k_rr_base<128, 32><<<dim3((uint32_t)blocks), dim3(256)>>>(
        nq, ncand, q, store, kin, kout);
  }, false);
  int64_t pairs = (total + 1) / 2;
  run("v1 ilp2", [&] {
    k_rr_ilp2<128><<<dim3((uint32_t)((pairs + 255) / 256)), dim3(256)>>>(
        nq, ncand, q, store, kin, kout);
  }, false);
  run("v2 lds BS256", [&] {
    int nchunk = (ncand + 255) / 256;
    k_rr_lds<128, 256><<<dim3((uint32_t)(nq * nchunk)), dim3(256)>>>(
        nq, ncand, q, store, kin, kout);
  }, false);
  run("v2 lds BS128", [&] {
    int nchunk = (ncand + 127) / 128;
    k_rr_lds<128, 128><<<dim3((uint32_t)(nq * nchunk)), dim3(128)>>>(
        nq, ncand, q, store, kin, kout);
  }, false);
  return 0;
}
