/*
 * select.hpp — device-side exact top-k selection on u64 (dist,id) keys.
 *
 * Key layout: upper 32 bits = monotone u32 image of the fp32 distance
 * (sign-flip trick; inverted for inner-product so ascending key ==
 * descending similarity), lower 32 bits = docid. One integer compare gives
 * the exact (dist, id) total order — ties break by id deterministically
 * (SURVEY §8c pin (i)); replaces the reference's faiss CMin/CMax heaps
 * (gamma_index_ivfpq.cc:606-631, gamma_index_flat.cc:80-83).
 *
 * GammaSelector: each thread keeps a register buffer of candidates better
 * than the current block-wide threshold; the buffers are periodically
 * dumped to LDS and bitonic-merged with the running top-k, which
 * re-tightens the threshold. Exact: a candidate is dropped only when it is
 * >= the k-th best key seen so far.
 *
 * Protocol: between two maybe_flush() calls, each thread may push() at
 * most GAMMA_SEL_CHUNK candidates. All threads of the block must reach
 * maybe_flush()/finish() together (they contain barriers).
 */
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define GAMMA_KEY_EMPTY 0xffffffffffffffffull

__device__ __forceinline__ uint32_t gamma_f32_key(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

__device__ __forceinline__ float gamma_key_f32(uint32_t dk) {
  uint32_t u = (dk & 0x80000000u) ? (dk & 0x7fffffffu) : ~dk;
  return __uint_as_float(u);
}

template <bool IP>
__device__ __forceinline__ uint64_t gamma_make_key(float dist, uint32_t id) {
  uint32_t dk = gamma_f32_key(dist);
  if (IP) dk = ~dk;
  return ((uint64_t)dk << 32) | (uint64_t)id;
}

template <bool IP>
__device__ __forceinline__ float gamma_key_dist(uint64_t key) {
  uint32_t dk = (uint32_t)(key >> 32);
  if (IP) dk = ~dk;
  return gamma_key_f32(dk);
}

__device__ __forceinline__ int64_t gamma_key_id(uint64_t key) {
  return (key == GAMMA_KEY_EMPTY) ? (int64_t)-1
                                  : (int64_t)(key & 0xffffffffull);
}

/* In-LDS bitonic sort (ascending) of buf[0..n), n a power of two.
 * Call with all threads of the block; barriers inside. */
__device__ inline void gamma_bitonic_sort(uint64_t *buf, int n) {
  const int tid = threadIdx.x, nt = blockDim.x;
  for (int len = 2; len <= n; len <<= 1) {
    for (int inc = len >> 1; inc > 0; inc >>= 1) {
      __syncthreads();
      for (int i = tid; i < n; i += nt) {
        int j = i ^ inc;
        if (j > i) {
          bool up = (i & len) == 0;
          uint64_t a = buf[i], b = buf[j];
          if ((a > b) == up) { buf[i] = b; buf[j] = a; }
        }
      }
    }
  }
  __syncthreads();
}

#define GAMMA_SORT_CAP 2048   /* LDS sort scratch, u64 x 2048 = 16 KB */
#define GAMMA_SEL_MAX_K 1088  /* cap - margin must stay >= 0 */

/* Candidates append into the shared LDS buffer through one LDS atomic;
 * all selector state is scalar registers or LDS (no per-thread arrays —
 * runtime-indexed register arrays land in scratch memory, §5.4 rule 20,
 * which made v1 of this selector 60x slower).
 *
 * Invariant: the caller calls maybe_flush(margin) (all threads together)
 * often enough that at most `margin` pushes can happen block-wide between
 * two checks; the buffer then never overflows its cap = SORT_CAP - k. */
struct GammaSelector {
  uint64_t *sortbuf;  /* uint64_t[GAMMA_SORT_CAP] in LDS */
  uint64_t *res;      /* uint64_t[k] in LDS, sorted ascending after flush */
  int *state;         /* int[1] in LDS: append counter */
  int k, cap;
  uint64_t thresh;

  __device__ void init(uint64_t *sortbuf_, uint64_t *res_, int *state_,
                       int k_) {
    sortbuf = sortbuf_; res = res_; state = state_; k = k_;
    cap = GAMMA_SORT_CAP - k;
    thresh = GAMMA_KEY_EMPTY;
    for (int i = threadIdx.x; i < k; i += blockDim.x) res[i] = GAMMA_KEY_EMPTY;
    if (threadIdx.x == 0) state[0] = 0;
    __syncthreads();
  }

  /* seed the running result with previously selected keys (unsorted ok) */
  __device__ void seed(const uint64_t *keys, int n) {
    for (int i = threadIdx.x; i < k; i += blockDim.x)
      res[i] = (i < n) ? keys[i] : GAMMA_KEY_EMPTY;
    __syncthreads();
    flush_();
  }

  __device__ __forceinline__ void push(uint64_t key) {
    if (key < thresh) {
      int idx = atomicAdd(&state[0], 1);
      sortbuf[idx] = key; /* in range by the maybe_flush invariant */
    }
  }

  /* all threads must arrive together */
  __device__ __forceinline__ void maybe_flush(int margin) {
    __syncthreads();
    if (state[0] > cap - margin) flush_();
  }

  __device__ void finish() {
    __syncthreads();
    flush_();
  }

 private:
  __device__ void flush_() {
    const int nt = blockDim.x, tid = threadIdx.x;
    int total = state[0];
    if (total > cap) total = cap; /* belt & braces */
    /* append res at [total, total+k), pad to pow2, sort, keep k */
    int n = 1;
    while (n < total + k) n <<= 1;
    __syncthreads();
    for (int i = total + k + tid; i < n; i += nt)
      sortbuf[i] = GAMMA_KEY_EMPTY;
    for (int i = tid; i < k; i += nt) sortbuf[total + i] = res[i];
    gamma_bitonic_sort(sortbuf, n);
    for (int i = tid; i < k; i += nt) res[i] = sortbuf[i];
    if (tid == 0) state[0] = 0;
    __syncthreads();
    thresh = res[k - 1];
  }
};

/* delete-bitmap test (1 = deleted), u32 words; null bitmap = none */
__device__ __forceinline__ bool gamma_bitmap_test(const uint32_t *bm,
                                                  uint64_t id) {
  return bm && ((bm[id >> 5] >> (id & 31)) & 1u);
}

/* ------------------------------------------------------------------------
 * Scale note: on the BIG ADC scan (10M codes, 275k candidates/query)
 * this wave selector validates bit-exact but runs 38-50% SLOWER than the
 * block selector (13.6-17.1 ms vs 9.1-9.8 ms, tools/adc_bench.hip): the
 * 64-lane bitonic flush does 8 elems/lane over 45 substages and fires
 * ~4x as often (cap 312 vs 1848). It wins where per-query data is SMALL
 * (k_select_from_dots_wave: top-nprobe of nlist=4096 coarse distances,
 * ~k2*ln(n/k2) pushes => ~1 flush total) because it drops every block
 * barrier and packs 8 queries per workgroup.
 *
 * Wave-local exact top-k selector: one 64-lane wave owns a private LDS
 * region, so the scan needs NO cross-wave barriers (a wave is lockstep;
 * __builtin_amdgcn_wave_barrier() pins the compiler's DS ordering at
 * phase boundaries). The block merges the per-wave results once at the
 * end.
 * Region layout per wave (buf = uint64_t[GAMMA_WSEL_CAP], pow2):
 *   [0, k)            running top-k, sorted ascending after a flush
 *   [k, GAMMA_WSEL_CAP) append region (cap = GAMMA_WSEL_CAP - k slots)
 * flush sorts [0, k+total) in place (res already in front — no copies,
 * no overlap) and re-tightens the threshold.
 * Invariant: maybe_flush(margin) runs at least every `margin`
 * wave-pushes and k + margin <= GAMMA_WSEL_CAP. */
#define GAMMA_WSEL_CAP 512

struct GammaWaveSelector {
  uint64_t *buf; /* uint64_t[GAMMA_WSEL_CAP] in LDS, per wave */
  int *cnt;      /* int[1] in LDS, per wave */
  int k, cap;
  uint64_t thresh;

  __device__ void init(uint64_t *buf_, int *cnt_, int k_) {
    buf = buf_; cnt = cnt_; k = k_;
    cap = GAMMA_WSEL_CAP - k;
    thresh = GAMMA_KEY_EMPTY;
    const int lane = threadIdx.x & 63;
    for (int i = lane; i < k; i += 64) buf[i] = GAMMA_KEY_EMPTY;
    if (lane == 0) *cnt = 0;
    __builtin_amdgcn_wave_barrier();
  }

  __device__ __forceinline__ void push(uint64_t key) {
    if (key < thresh) {
      int idx = atomicAdd(cnt, 1);
      buf[k + idx] = key; /* in range by the maybe_flush invariant */
    }
  }

  __device__ __forceinline__ void maybe_flush(int margin) {
    __builtin_amdgcn_wave_barrier();
    if (*cnt > cap - margin) flush_();
  }

  __device__ void finish() {
    __builtin_amdgcn_wave_barrier();
    flush_();
  }

  /* wave-level bitonic sort of [0, k+total): lockstep lanes,
   * wave_barrier pins DS ordering between stages */
  __device__ void flush_() {
    const int lane = threadIdx.x & 63;
    int total = *cnt;
    if (total > cap) total = cap;
    int n = 1;
    while (n < k + total) n <<= 1; /* n <= GAMMA_WSEL_CAP (pow2) */
    __builtin_amdgcn_wave_barrier();
    for (int i = k + total + lane; i < n; i += 64)
      buf[i] = GAMMA_KEY_EMPTY;
    __builtin_amdgcn_wave_barrier();
    for (int len = 2; len <= n; len <<= 1) {
      for (int inc = len >> 1; inc > 0; inc >>= 1) {
        for (int i = lane; i < n; i += 64) {
          int j = i ^ inc;
          if (j > i) {
            bool up = (i & len) == 0;
            uint64_t a = buf[i], b = buf[j];
            if ((a > b) == up) { buf[i] = b; buf[j] = a; }
          }
        }
        __builtin_amdgcn_wave_barrier();
      }
    }
    if (lane == 0) *cnt = 0;
    __builtin_amdgcn_wave_barrier();
    thresh = buf[k - 1];
  }
};
