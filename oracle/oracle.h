/* oracle.h — CPU restatement of dingo-store's vector-search hot path.
 *
 * TEST INFRASTRUCTURE ONLY.  This library is the parity oracle and the CPU
 * baseline timer.  Only tests/, __graft_entry__.smoke() and
 * bench.py's cpu_baseline leg may import, call, link or execute it — never
 * the product path (dingo-store_amd/), which must fail loudly without its
 * HIP extension.
 *
 * Parity pinning status: the arithmetic core (dgo_fvec_L2sqr /
 * dgo_fvec_inner_product / dgo_fvec_norm_L2sqr, and the normalize rule) is
 * pinned BIT-EXACT against the reference's own in-tree scalar code,
 * /root/reference/src/simd/distances_ref.cc:24-64 compiled unmodified into
 * oracle/_ref/libdistref.so (tests/test_oracle.py).  The index-level
 * semantics (IVF parameter handling, result shaping, untrained behavior)
 * are restated from the reference call sites cited per function below.  The
 * faiss index internals the reference delegates to live in the EMPTY
 * submodule contrib/faiss (fork dingodb/faiss, pin unrecoverable —
 * /root/reference/.gitmodules:13-15); their published
 * algorithms (IndexFlat scan, IndexIVFFlat raw-vector lists, Clustering
 * k-means defaults niter=25 / max 256 points per centroid / seed 1234) are
 * restated here and validated by the property tests the reference itself
 * pins (Flat self-top-1, test_vector_index_recall_flat.cc:170-236) plus
 * recall-vs-exhaustive-ground-truth properties the reference lacks.
 */
#ifndef DGO_ORACLE_H_
#define DGO_ORACLE_H_
#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* metrics mirror dg_metric in include/dingo_gpu.h */
enum { DGO_L2 = 0, DGO_IP = 1, DGO_COSINE = 2 };

/* --- arithmetic core: bit-exact restatement of src/simd/distances_ref.cc --- */
float dgo_fvec_L2sqr(const float* x, const float* y, size_t d);        /* :24-31 */
float dgo_fvec_inner_product(const float* x, const float* y, size_t d);/* :53-60 */
float dgo_fvec_norm_L2sqr(const float* x, size_t d);                   /* :62-66 */

/* NormalizeVectorForFaiss, src/vector/vector_index_utils.cc:480-491 */
void dgo_normalize(float* x, int32_t d);
void dgo_normalize_batch(float* x, int64_t n, int32_t d);

/* --- Flat exhaustive search (faiss IndexFlatL2/IP via IndexIDMap2;
 * reference VectorIndexFlat::Search, src/vector/vector_index_flat.cc:205-264).
 * Distances in faiss convention (L2 raw sqr; IP raw score, larger better).
 * out_ids padded -1; ties broken toward smaller id.  queries are used as
 * given (caller normalizes for cosine, as ExtractVectorValue does).
 * ids == NULL => implicit ids 0..n-1. --- */
void dgo_flat_search(int metric, int64_t n, int32_t d, const float* base,
                     const int64_t* ids, int64_t nq, const float* queries,
                     int32_t k, float* out_dist, int64_t* out_ids);

/* --- k-means (faiss Clustering restated: niter=25, seed=1234, subsample to
 * 256*nlist points, empty-cluster split; assignment by `metric`).
 * Invoked by reference Train, src/vector/vector_index_ivf_flat.cc:644-712
 * (degrade nlist->1 when n < nlist happens in the CALLER / shim, :676-680).
 * centroids_out: nlist x d. --- */
void dgo_kmeans(int metric, int64_t n, int32_t d, const float* x,
                int32_t nlist, int32_t niter, uint32_t seed,
                float* centroids_out);

/* coarse assignment: nearest centroid per vector by metric (faiss
 * IndexIVF::add_with_ids preamble; ivf_flat.cc:119-124) */
void dgo_ivf_assign(int metric, int64_t n, int32_t d, const float* x,
                    int32_t nlist, const float* centroids, int32_t* assign_out);

/* Build CSR inverted lists from assignments: offsets[nlist+1]; vectors/ids
 * permuted into list-grouped order, arrival order preserved within a list
 * (faiss InvertedLists append order). */
void dgo_ivf_build(int64_t n, int32_t d, const float* x, const int64_t* ids,
                   int32_t nlist, const int32_t* assign, int64_t* offsets_out,
                   float* grouped_vectors_out, int64_t* grouped_ids_out);

/* --- IVF-Flat search (faiss IndexIVFFlat::search restated; reference
 * VectorIndexIvfFlat::Search src/vector/vector_index_ivf_flat.cc:191-275:
 * nprobe pre-clamped by caller to [1, nlist]).  list_mask: optional
 * per-list ownership mask (NULL = all); mirrors multi-GPU sharding. --- */
void dgo_ivf_search(int metric, int32_t nlist, int32_t d,
                    const float* centroids, const int64_t* offsets,
                    const float* grouped_vectors, const int64_t* grouped_ids,
                    int64_t nq, const float* queries, int32_t k,
                    int32_t nprobe, const uint8_t* list_mask,
                    float* out_dist, int64_t* out_ids);

/* coarse top-nprobe per query (exposed for tests): probe list ids in rank
 * order, -1 padded. */
void dgo_coarse_probe(int metric, int32_t nlist, int32_t d,
                      const float* centroids, int64_t nq,
                      const float* queries, int32_t nprobe,
                      int32_t* probes_out);

/* --- IVF-PQ (faiss IndexIVFPQ restated: residual encoding, per-subspace
 * 256-centroid codebooks, ADC scan; reference VectorIndexRawIvfPq::Search
 * src/vector/vector_index_raw_ivf_pq.cc:157-210). --- */
void dgo_pq_train(int64_t n, int32_t d, const float* residuals, int32_t m,
                  int32_t nbits, uint32_t seed, float* codebooks_out);
void dgo_ivfpq_encode(int64_t n, int32_t d, const float* x,
                      const int32_t* assign, const float* centroids,
                      int32_t m, const float* codebooks, uint8_t* codes_out);
void dgo_ivfpq_search(int metric, int32_t nlist, int32_t d,
                      const float* centroids, const int64_t* offsets,
                      const uint8_t* grouped_codes, const int64_t* grouped_ids,
                      int32_t m, const float* codebooks, int64_t nq,
                      const float* queries, int32_t k, int32_t nprobe,
                      float* out_dist, int64_t* out_ids);

/* --- vectorized (reordered-sum) variants for bench.py's cpu_baseline leg
 * ONLY (oracle_fast.c; the reference's hot loops are AVX512 via src/simd
 * hooks, so the timed CPU baseline uses these; parity uses the strict
 * versions above). --- */
void dgo_flat_search_fast(int metric, int64_t n, int32_t d, const float* base,
                          const int64_t* ids, int64_t nq, const float* queries,
                          int32_t k, float* out_dist, int64_t* out_ids);
void dgo_ivf_search_indexed_fast(int metric, int32_t nlist, int32_t d,
                                 const float* centroids,
                                 const int64_t* member_offsets,
                                 const int64_t* member_rows,
                                 const float* base, int64_t nq,
                                 const float* queries, int32_t k,
                                 int32_t nprobe, float* out_dist,
                                 int64_t* out_ids);
void dgo_ivf_search_fast(int metric, int32_t nlist, int32_t d,
                         const float* centroids, const int64_t* offsets,
                         const float* grouped_vectors,
                         const int64_t* grouped_ids, int64_t nq,
                         const float* queries, int32_t k, int32_t nprobe,
                         const uint8_t* list_mask, float* out_dist,
                         int64_t* out_ids);


/* --- range search (faiss RangeSearch restated; results best-first, ties
 * toward smaller id; lims always filled, entries beyond cap dropped) --- */
void dgo_flat_range_search(int metric, int64_t n, int32_t d,
                           const float* base, const int64_t* ids, int64_t nq,
                           const float* queries, float radius, int64_t* lims,
                           int64_t cap, float* out_dist, int64_t* out_ids);
void dgo_ivf_range_search(int metric, int32_t nlist, int32_t d,
                          const float* centroids, const int64_t* offsets,
                          const float* grouped_vectors,
                          const int64_t* grouped_ids, int64_t nq,
                          const float* queries, float radius, int32_t nprobe,
                          int64_t* lims, int64_t cap, float* out_dist,
                          int64_t* out_ids);

const char* dgo_version(void);

#ifdef __cplusplus
}
#endif
#endif
