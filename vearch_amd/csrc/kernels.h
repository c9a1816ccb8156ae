/* kernels.h — host API of the gfx950 HIP kernels (kernels.hip). */
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

/* One IVF bucket as the device sees it (RT list storage, SURVEY §8 a7;
 * semantics of realtime_mem_data.cc:57-68: append-only SoA with a
 * per-entry delete mark). Device ids are compressed to u32 with bit 31
 * as the delete mark (docids < 2^31 per partition); every host-visible
 * surface (debug API, Dump format, oracle) keeps the reference's int64 +
 * bit-63 form — SURVEY §8d notes the 4-byte id option and the roofline
 * stays on the algorithmic 8-byte ids. For IVFPQ `data` is uint8 codes
 * (code_size per entry); for IVFFLAT fp32 vectors (d floats). */
struct GammaBucketDev {
  const uint32_t *ids;
  const void *data;
  /* IVFPQ only: per-entry S term, S_v = sum_m btab[ln][m][code_m]
   * (the list-dependent half of the decomposed ADC tables, folded into
   * one float per vector at encode time so the scan never touches the
   * B table — see kernels.h pq_sterm). Null for IVFFLAT. */
  const float *svals;
  long long size;
};

/* one bucket's slice of a bulk-ingest batch (gk::bucket_scatter):
 * copy `count` entries starting at src_start of the staged arrays into
 * this bucket's device arrays at the recorded append position */
struct GammaScatterSeg {
  uint32_t *ids_dst;
  uint8_t *data_dst;
  float *sval_dst; /* null for IVFFLAT */
  long long src_start;
  long long count;
};

namespace gk {

/* Bulk bucket append: instead of 3 small hipMemcpys per bucket per
 * chunk (~100k calls per 64k-vector chunk at nlist=32k — the N=50M
 * build-time killer), the host stages the grouped chunk once and one
 * kernel scatters it into every bucket. */
hipError_t bucket_scatter(hipStream_t s, int nseg,
                          const GammaScatterSeg *segs_dev,
                          const uint32_t *ids_src,
                          const uint8_t *data_src,
                          const float *svals_src, int entry_bytes);

/* out[i*n+j] = dot(Q_i, B_j); f32 MFMA 16x16x4. */
hipError_t dots_mfma(hipStream_t s, const float *Q, int nq, const float *B,
                     int64_t n, int d, float *out);

/* norms[i] = sum_j v[i][j]^2, canonical sequential fmaf. */
hipError_t row_norms(hipStream_t s, const float *v, int64_t n, int d,
                     float *norms);

/* Exact top-k2 select per row of a dots matrix.
 * dist = l2 ? (qn[q] + bn[col_base+c] - 2*dot) : dot; id = col_base + c.
 * bitmap: optional delete bitmap over global ids (u32 words).
 * state_keys (nq x k2 u64) is seeded from itself when `seeded`. */
hipError_t select_from_dots(hipStream_t s, int nq, int64_t ncols,
                            int64_t col_base, int64_t ld, const float *dots,
                            const float *qnorms, const float *bnorms,
                            bool l2, bool ip_order, const uint32_t *bitmap,
                            int k2, uint64_t *state_keys, bool seeded);

/* Full-sort row select for short rows (ncols <= 8192, no bitmap):
 * one workgroup bitonic-sorts the whole row of (dist,id) keys and emits
 * the top-k2 as (dists, ids) — used for the coarse top-nprobe
 * (quantizer->search, ivfpq.cc:595). Same key order as
 * select_from_dots. */
hipError_t select_rows_full(hipStream_t s, int nq, int ncols, int64_t ld,
                            const float *dots, const float *qnorms,
                            const float *bnorms, bool l2, bool ip_order,
                            int k2, float *out_dists, int64_t *out_ids);

/* argmin over each row (ties -> lowest col). out int32[nrows]. */
hipError_t argmin_rows(hipStream_t s, int64_t nrows, int ncols,
                       const float *dots, const float *qnorms,
                       const float *bnorms, bool l2, int32_t *out);

/* Decomposed ADC tables (use_precomputed_table=1 semantics,
 * gamma_index_ivfpq.h:254-262):
 *   btab[ln][m][j] = 2 * (c_ln,m . cw_mj)        (train-time)
 *   atab[q][m][j]  = fmaf(-2, q_m . cw, ||cw||^2) (per search batch)
 * Sequential-fmaf arithmetic matching oracle_pct1_*_table. */
hipError_t pq_tables_b(hipStream_t s, int d, int M, int nlist,
                       const float *centroids, const float *codebooks,
                       float *btab);
hipError_t pq_tables_a(hipStream_t s, int nq, int d, int M,
                       const float *queries, const float *codebooks,
                       float *atab);

/* Per-vector S term (encode/add/load time):
 *   out[i] = sum_m btab[asg_i][m][code_i[m]]   (plain adds, m order)
 * where asg_i = asg ? asg[i] : asg_const. Folding the list half of the
 * pct1 decomposition into one float per vector removes the per-(query,
 * list) B-table staging (nprobe x M x ksub floats per query) from the
 * scan: dis = coarse_dis + S_v + sum_m atab[q][m][code_m], which is
 * the same T = A + B sum grouped per vector — fp32-rounding-only
 * difference, same class as the documented pct1 deviation
 * (DESIGN.md "The L2 ADC table mode"). */
hipError_t pq_sterm(hipStream_t s, int64_t n, int M, int nlist,
                    const uint8_t *codes, const int32_t *asg,
                    int asg_const, const float *btab, float *out);
/* Load-path batch: one launch rebuilds every bucket's S terms */
hipError_t pq_sterm_buckets(hipStream_t s, int nlist, int M,
                            const GammaBucketDev *bks, const float *btab);

/* IVFPQ fused search: one workgroup per query; stages the query-level
 * table ONCE (L2: A_q from atab, dis = coarse_dis + S_v + sum A[c_m];
 * IP: the h:164-167 query table, dis = dis0 + sum T[c_m]), then scans
 * the probed lists (h:923-953 semantics). out_keys: nq x k2. */
/* S = probe-split factor: S sub-workgroups per query (out_keys is
 * nq x S x k2; merge with sort_rows). S>1 serves small batches. */
/* qmap: optional query schedule — workgroup b serves query
 * qmap[b/S]. The engine sorts large batches by their first probed
 * list so neighbouring workgroups scan the same lists while they are
 * L2/LLC-resident (the 10M-doc code store is ~400 MB against the
 * 256 MiB Infinity Cache — unordered queries re-fetch every list
 * ~nq*nprobe/nlist times from DRAM). Results are identical: the
 * selector's top-k2 is order-independent and outputs are written at
 * the ORIGINAL query row. */
hipError_t ivfpq_scan(hipStream_t s, int nq, int S, int d, int M,
                      int nprobe,
                      int k2, const float *queries, const float *centroids,
                      const float *codebooks, const float *atab,
                      const float *probe_dists,
                      const GammaBucketDev *buckets,
                      int nlist, const int64_t *probes,
                      const uint32_t *bitmap, bool ip, uint64_t *out_keys,
                      const int *kill_flag, const int32_t *qmap);

/* out[q] = (int32) probes[q*nprobe] (the schedule sort key) */
hipError_t extract_probe0(hipStream_t s, int nq, int nprobe,
                          const int64_t *probes, int32_t *out);

/* IVFFLAT fused search (gamma_index_ivfflat.h:36-91). */
hipError_t ivfflat_scan(hipStream_t s, int nq, int d, int nprobe, int k2,
                        const float *queries, const GammaBucketDev *buckets,
                        int nlist, const int64_t *probes,
                        const uint32_t *bitmap, bool ip, uint64_t *out_keys);

/* Canonical-order exact re-rank (ivfpq.cc:675-726 + parity canonicalizer):
 * for each key in keys_in (nq x ncand), gather the raw vector through the
 * segment table and recompute the distance with the sequential fmaf loop.
 * keys_out may alias keys_in. */
hipError_t rerank(hipStream_t s, int nq, int ncand, int d,
                  const float *queries, const float *const *segs,
                  int n_segs, int seg_shift, bool ip,
                  const uint64_t *keys_in, uint64_t *keys_out);

/* Per-row sort of ncand keys (<= 2048), emit top-k (dists, ids). */
hipError_t sort_rows(hipStream_t s, int nq, int ncand, int k,
                     const uint64_t *keys, bool ip, float *out_dists,
                     int64_t *out_ids);

/* unpack keys -> (dists, ids) without sorting (keys already sorted). */
hipError_t unpack_keys(hipStream_t s, int64_t n, const uint64_t *keys,
                       bool ip, float *out_dists, int64_t *out_ids);

/* residuals[i] = x[i] - centroids[assign[i]] (elementwise exact). */
hipError_t residuals(hipStream_t s, int64_t n, int d, const float *x,
                     const float *centroids, const int32_t *assign,
                     float *out);

/* PQ encode: codes[i][m] = argmin_j ||x_sub - codebook[m][j]||^2
 * (pq.compute_codes, ivfpq.cc:494; ties -> lowest j). */
hipError_t pq_encode(hipStream_t s, int64_t n, int d, int M, int ksub,
                     const float *x, const float *codebooks, uint8_t *codes);

/* FLAT brute-force over the raw-vector segment table (no materialized
 * dist matrix): per query workgroup streams all n vectors. Used for small
 * nq; large nq goes through dots_mfma chunks + select_from_dots. */
hipError_t flat_stream_scan(hipStream_t s, int nq, int64_t n, int d, int k2,
                            const float *queries, const float *const *segs,
                            int seg_shift, const uint32_t *bitmap, bool ip,
                            uint64_t *out_keys);

}  // namespace gk
