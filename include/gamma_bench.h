/*
 * gamma_bench.h — auxiliary (non-reference) entry points of libgamma.so.
 *
 * These symbols are NOT part of the reference C ABI (include/gamma_api.h);
 * they exist for the test/bench harness only:
 *  - bulk ingest that is semantically equivalent to a loop of
 *    AddOrUpdateDoc (reference: gamma_api.h:52) without per-doc FlatBuffers
 *    marshalling, so a 10M-vector index can be built in seconds;
 *  - a raw batched search equal to the compute path behind Search
 *    (reference: gamma_api.cc:175 -> engine.cc:248 -> vector_manager.cc:851)
 *    without protobuf marshalling, so the bench can time the hot path with
 *    queries already resident in HBM (the protobuf-inclusive rate is
 *    reported separately in DESIGN.md);
 *  - layered debug hooks used by the GPU parity tests to pin each stage of
 *    the IVFPQ pipeline against the CPU oracle.
 */

#ifndef GAMMA_BENCH_H_
#define GAMMA_BENCH_H_

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Bulk-add n vectors (row-major float32 n*d) for vector field `field`.
 * Docids are assigned sequentially from the current max docid; the i-th
 * vector's primary key is its docid rendered as a decimal string.
 * Returns 0 on success. */
int GammaBulkAdd(void *engine, const char *field, int field_len, int n,
                 const float *vecs);

/* Batched search on the single vector index of `engine`.
 * xq: nq*d float32 (host). k results per query.
 * nprobe<=0 -> index default. rerank>0 -> exact re-rank of `rerank`
 * candidates (reference recall_num, ivfpq.cc:675-726).
 * metric: 0 = engine default, 1 = L2, 2 = InnerProduct.
 * out_dists: nq*k float32; out_ids: nq*k int64 (-1 padded).
 * Returns 0 ok, -2 killed. */
int GammaRawSearch(void *engine, int nq, const float *xq, int k, int nprobe,
                   int rerank, int metric, float *out_dists,
                   int64_t *out_ids);

/* Upload nq*d queries once into HBM; GammaRawSearchCached then times the
 * hot path with inputs already device-resident (bench contract). */
int GammaCacheQueries(void *engine, int nq, const float *xq);
int GammaRawSearchCached(void *engine, int nq, int k, int nprobe,
                         int rerank, int metric, float *out_dists,
                         int64_t *out_ids);

/* Debug: run only the coarse-assign stage (quantizer->search equivalent,
 * reference ivfpq.cc:595): top-nprobe centroids per query. */
int GammaDebugCoarseAssign(void *engine, int nq, const float *xq, int nprobe,
                           int64_t *out_lists, float *out_dists);

/* Debug: copy the trained model to host: centroids (nlist*d), pq codebooks
 * (M*ksub*dsub). Buffers may be NULL to skip. */
int GammaDebugGetModel(void *engine, float *centroids, float *codebooks);
/* OPQ debug hooks (oracle parity chains on the engine's own R and its
 * GPU rotation so downstream ADC stays bit-exact):
 * GetOPQ copies the d*d row-major rotation; ApplyOPQ returns the
 * engine-rotated queries (nq*d in, nq*d out). -1 if no OPQ. */
int GammaDebugGetOPQ(void *engine, float *R);
int GammaDebugApplyOPQ(void *engine, const float *xq, int nq, float *out);

/* Debug: fetch the contents of inverted list `list_no`: returns size, and
 * copies ids (int64, with bit-63 delete marks preserved, reference
 * realtime_mem_data.h:26) and codes (size*code_size u8) if non-NULL. */
int64_t GammaDebugGetList(void *engine, int64_t list_no, int64_t *ids,
                          uint8_t *codes);

/* Number of indexed vectors / engine docs. */
int64_t GammaDebugNumDocs(void *engine);

/* Timing of the last GammaRawSearch, in microseconds, split by stage:
 * [0]=H2D, [1]=coarse assign+select, [2]=list scan (the dominant kernel),
 * [3]=rerank+merge, [4]=D2H, [5]=total wall. Returns 0. */
int GammaLastSearchTiming(void *engine, double *us6);

#ifdef __cplusplus
}
#endif

#endif /* GAMMA_BENCH_H_ */
