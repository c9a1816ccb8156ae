/* ref_harness.cpp — executes the REFERENCE'S OWN scanner code (test
 * infrastructure ONLY; nothing in the product path links or loads this
 * library — see oracle/ref_scan.c's header for the oracle usage rules).
 *
 * The extracted .inc files under _ref/gen/ are produced at build time
 * by oracle/ref_extract.sh from /root/reference (verbatim reference
 * lines; gitignored). This file provides the drivers around them,
 * restating only the control flow of the reference call sites it
 * cites:
 *   - ref_flat_search:   gamma_index_flat.cc:366-513 Search
 *     (OMP over queries :452, heapify/scan/reorder :455-468)
 *   - ref_ivfpq_search:  gamma_index_ivfpq.cc:730-933
 *     search_preassigned (OMP :800, per query set_query :816,
 *     scan_one_list :635 -> set_list + scan_codes; the
 *     precompute_mode==2 branch of scan_codes h:968-970 calls the
 *     extracted scan_list_with_table)
 *   - ref_ivfflat_search: gamma_index_ivfflat.cc:579-770
 *     search_preassigned driving GammaIVFFlatScanner::scan_codes
 * Coarse assignment (quantizer->search, ivfpq.cc:595) is an input:
 * callers pass the probed list ids + coarse distances, so the scan
 * parity is independent of how the probes were produced.
 */
#include "ref_stub.hpp"

/* reference lines, extracted verbatim at build time: */
#include "_ref/gen/qtables.inc"    /* QueryTables + WrappedSearchResult */
#include "_ref/gen/flat_scan.inc"  /* FlatScanCtx + ComputeScoreBatch +
                                      FlatScanRange */
#include "_ref/gen/ivfflat_scanner.inc" /* GammaIVFFlatScanner */

/* The Gamma IVFPQ scanner: QueryTables (extracted) + the extracted
 * dis0/init_list and scan_list_with_table members, wired exactly as
 * GammaIVFPQScanner does (gamma_index_ivfpq.h:783-953: set_query ->
 * init_query h:809, set_list -> init_list(.., precompute_mode) h:813,
 * precompute_mode = 2 for the table scan). */
template <faiss::MetricType METRIC_TYPE, class C>
struct RefGammaIVFPQScanner : QueryTables {
  using PQDecoder = faiss::PQDecoder8;
  const RetrievalContext *retrieval_context_;

  RefGammaIVFPQScanner(const faiss::IndexIVFPQ &ivfpq,
                       const RetrievalContext *rc)
      : QueryTables(ivfpq, nullptr, METRIC_TYPE), retrieval_context_(rc) {}

#include "_ref/gen/ivfpq_init_list.inc" /* float dis0; init_list() */
#include "_ref/gen/gamma_scan.inc"      /* scan_list_with_table() */

  void set_query(const float *query) { this->init_query(query); }
  void set_list(idx_t list_no, float coarse_dis) {
    this->init_list(list_no, coarse_dis, /*precompute_mode=*/2);
  }
};

extern "C" {

/* FLAT exact scan, L2 (metric_ip=0) / IP (1). del_bitmap: 1 bit per
 * vid, 1 = deleted, may be null. */
int ref_flat_search(int64_t n, int d, const float *base,
                    const uint8_t *del_bitmap, int nq, const float *q,
                    int k, int metric_ip, float *out_d, int64_t *out_i) {
  try {
    RetrievalContext rc{del_bitmap};
    RawVector rv{base, d};
    static const std::string kRid;
#pragma omp parallel for schedule(dynamic)
    for (int i = 0; i < nq; i++) {
      FlatScanCtx ctx{&rc, &rv, d, k, 1024, kRid, 0};
      float *simi = out_d + (size_t)i * k;
      idx_t *idxi = out_i + (size_t)i * k;
      if (metric_ip) {
        faiss::heap_heapify<HeapForIP>(k, simi, idxi);
        FlatScanRange<HeapForIP>(ctx, q + (size_t)i * d, 0, (int)n, simi,
                                 idxi);
        faiss::heap_reorder<HeapForIP>(k, simi, idxi);
      } else {
        faiss::heap_heapify<HeapForL2>(k, simi, idxi);
        FlatScanRange<HeapForL2>(ctx, q + (size_t)i * d, 0, (int)n, simi,
                                 idxi);
        faiss::heap_reorder<HeapForL2>(k, simi, idxi);
      }
    }
    return 0;
  } catch (...) {
    return -1;
  }
}

} /* extern "C" (templates below cannot carry C linkage) */

static faiss::IndexIVFPQ *make_ivfpq(int d, int nlist, int M,
                                     const float *centroids,
                                     const float *codebooks, int upt,
                                     int metric_ip) {
  auto *quant = new faiss::IndexFlat();
  quant->d = d;
  quant->xb.assign(centroids, centroids + (size_t)nlist * d);
  auto *ix = new faiss::IndexIVFPQ();
  ix->d = d;
  ix->pq.init(d, M);
  memcpy(ix->pq.centroids.data(), codebooks,
         ix->pq.centroids.size() * sizeof(float));
  ix->quantizer = quant;
  ix->by_residual = true; /* ivfpq.cc:195 */
  ix->use_precomputed_table = upt;
  if (upt == 1 && !metric_ip) ix->precompute_table();
  return ix;
}

template <faiss::MetricType MT, class C>
static void ivfpq_run(const faiss::IndexIVFPQ &ix,
                      const int64_t *list_offsets, const int64_t *ids,
                      const uint8_t *codes, int nq, const float *q, int d,
                      int M, int nprobe, const int64_t *probes,
                      const float *probe_dis, int k,
                      const uint8_t *del_bitmap, float *out_d,
                      int64_t *out_i) {
  RetrievalContext rc{del_bitmap};
#pragma omp parallel for schedule(dynamic)
  for (int qi = 0; qi < nq; qi++) {
    RefGammaIVFPQScanner<MT, C> scanner(ix, &rc);
    scanner.set_query(q + (size_t)qi * d);
    faiss::HeapResultHandler<C> handler(k, out_d + (size_t)qi * k,
                                        out_i + (size_t)qi * k);
    for (int p = 0; p < nprobe; p++) {
      idx_t ln = probes[(size_t)qi * nprobe + p];
      if (ln < 0) continue; /* scan_one_list ivfpq.cc:639 */
      scanner.set_list(ln, probe_dis[(size_t)qi * nprobe + p]);
      size_t lsz = list_offsets[ln + 1] - list_offsets[ln];
      WrappedSearchResult<C, false> res(ln, ids + list_offsets[ln],
                                        nullptr, handler);
      scanner.scan_list_with_table(
          lsz, codes + (size_t)list_offsets[ln] * M, res);
    }
    handler.end();
  }
}

/* IVFPQ ADC scan through the reference QueryTables + Gamma scan loop.
 * Lists in CSR form: list_offsets[nlist+1]; ids may carry bit 63 as
 * the delete mark (realtime_mem_data.h:26). use_precomputed_table: 0 =
 * the reference's runtime default (ivfpq.cc:196, per-(q,list) residual
 * tables), 1 = the documented decomposed mode (ivfpq.h:254-262, the
 * mode this repo's engine runs). */
extern "C" int ref_ivfpq_search(int d, int nlist, int M, const float *centroids,
                     const float *codebooks, const int64_t *list_offsets,
                     const int64_t *ids, const uint8_t *codes, int nq,
                     const float *q, int nprobe, const int64_t *probes,
                     const float *probe_dis, int metric_ip,
                     int use_precomputed_table, int k,
                     const uint8_t *del_bitmap, float *out_d,
                     int64_t *out_i) {
  try {
    faiss::IndexIVFPQ *ix =
        make_ivfpq(d, nlist, M, centroids, codebooks,
                   use_precomputed_table, metric_ip);
    if (metric_ip)
      ivfpq_run<faiss::METRIC_INNER_PRODUCT, faiss::CMin<float, idx_t>>(
          *ix, list_offsets, ids, codes, nq, q, d, M, nprobe, probes,
          probe_dis, k, del_bitmap, out_d, out_i);
    else
      ivfpq_run<faiss::METRIC_L2, faiss::CMax<float, idx_t>>(
          *ix, list_offsets, ids, codes, nq, q, d, M, nprobe, probes,
          probe_dis, k, del_bitmap, out_d, out_i);
    delete ix->quantizer;
    delete ix;
    return 0;
  } catch (...) {
    return -1;
  }
}

/* expose the mode-1 table build so tests can pin the decomposition
 * inputs themselves */
extern "C" int ref_ivfpq_precomputed_table(int d, int nlist, int M,
                                const float *centroids,
                                const float *codebooks, float *out) {
  try {
    faiss::IndexIVFPQ *ix =
        make_ivfpq(d, nlist, M, centroids, codebooks, 1, 0);
    memcpy(out, ix->precomputed_table.data(),
           ix->precomputed_table.size() * sizeof(float));
    delete ix->quantizer;
    delete ix;
    return 0;
  } catch (...) {
    return -1;
  }
}

template <faiss::MetricType MT, class C>
static void ivfflat_run(int d, const int64_t *list_offsets,
                        const int64_t *ids, const float *vecs, int nq,
                        const float *q, int nprobe, const int64_t *probes,
                        int k, const uint8_t *del_bitmap, float *out_d,
                        int64_t *out_i) {
  RetrievalContext rc{del_bitmap};
#pragma omp parallel for schedule(dynamic)
  for (int qi = 0; qi < nq; qi++) {
    GammaIVFFlatScanner<MT, C, false> scanner(d, false, nullptr, &rc);
    scanner.set_query(q + (size_t)qi * d);
    float *simi = out_d + (size_t)qi * k;
    idx_t *idxi = out_i + (size_t)qi * k;
    faiss::heap_heapify<C>(k, simi, idxi);
    for (int p = 0; p < nprobe; p++) {
      idx_t ln = probes[(size_t)qi * nprobe + p];
      if (ln < 0) continue;
      scanner.set_list(ln, 0.0f);
      size_t lsz = list_offsets[ln + 1] - list_offsets[ln];
      scanner.scan_codes(
          lsz, (const uint8_t *)(vecs + (size_t)list_offsets[ln] * d),
          ids + list_offsets[ln], simi, idxi, k);
    }
    faiss::heap_reorder<C>(k, simi, idxi);
  }
}

/* IVFFLAT scan through the extracted GammaIVFFlatScanner. */
extern "C" int ref_ivfflat_search(int d, int nlist, const int64_t *list_offsets,
                       const int64_t *ids, const float *vecs, int nq,
                       const float *q, int nprobe, const int64_t *probes,
                       int metric_ip, int k, const uint8_t *del_bitmap,
                       float *out_d, int64_t *out_i) {
  (void)nlist;
  try {
    if (metric_ip)
      ivfflat_run<faiss::METRIC_INNER_PRODUCT, faiss::CMin<float, idx_t>>(
          d, list_offsets, ids, vecs, nq, q, nprobe, probes, k,
          del_bitmap, out_d, out_i);
    else
      ivfflat_run<faiss::METRIC_L2, faiss::CMax<float, idx_t>>(
          d, list_offsets, ids, vecs, nq, q, nprobe, probes, k,
          del_bitmap, out_d, out_i);
    return 0;
  } catch (...) {
    return -1;
  }
}


