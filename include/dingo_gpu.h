/* dingo_gpu.h — C-ABI of the MI355X-native vector-index search path for
 * dingo-store's Index role.
 *
 * This is the drop-in boundary defined in SURVEY.md §8(b): the set of entry
 * points a replacement of the reference's VectorIndex plugin surface
 * (reference: src/vector/vector_index.h:56-279, concrete indexes
 * src/vector/vector_index_{flat,ivf_flat,raw_ivf_pq}.cc) must export.  The
 * C++ host mirror in dingo-store_amd/host/ re-declares the reference's
 * plugin virtuals (Search/RangeSearch/Add/Upsert/Delete/Train/Save/Load,
 * vector_index.h:148-229) on top of exactly these functions; a dingo-store
 * maintainer binds them from the Index role as shown in INTEGRATION.md.
 *
 * Conventions (mirroring the reference call sites):
 *  - Caller owns every in/out buffer except range-search results (dg_free).
 *  - No exceptions cross the ABI: int status + dg_last_error().
 *  - Distances are returned in faiss convention (L2: raw squared distance,
 *    IP/cosine: raw score, larger = better).  The observable dingo-store
 *    semantics — IP/cosine reported as 1.0f - score, L2 passed through —
 *    are applied by the C++ shim, restating
 *    src/vector/vector_index_utils.cc:612-655 (the flip at :634).
 *  - out_ids is padded with -1 where fewer than k results exist, like the
 *    labels pre-fill at src/vector/vector_index_flat.cc:218 and
 *    src/vector/vector_index_ivf_flat.cc:217.
 *  - Thread-safety: concurrent dg_search calls are safe; dg_add / dg_remove /
 *    dg_train / dg_load are exclusive (host RW lock mirroring the
 *    reference's RWLock usage, src/vector/vector_index_ivf_flat.cc:109,225).
 */
#ifndef DINGO_GPU_H_
#define DINGO_GPU_H_

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum dg_status {
  DG_OK = 0,
  DG_EINVAL = 1,        /* pb::error::EILLEGAL_PARAMTETERS */
  DG_ENOT_TRAINED = 2,  /* pb::error::EVECTOR_NOT_TRAIN */
  DG_ENOT_SUPPORT = 3,  /* pb::error::EVECTOR_NOT_SUPPORT — triggers the
                           reader's brute-force fallback,
                           src/vector/vector_reader.cc:1828-1831 */
  DG_ENOMEM = 4,
  DG_EINTERNAL = 5,     /* pb::error::EINTERNAL */
  DG_EIO = 6,
  DG_ENOGPU = 7,        /* HIP device unavailable: the product path fails
                           loudly, it never falls back to CPU */
  DG_EID_DUPLICATED = 8,/* pb::error::EVECTOR_ID_DUPLICATED */
  DG_ENOT_FOUND = 9     /* pb::error::EVECTOR_INVALID (remove miss),
                           src/vector/vector_index_ivf_flat.cc:183 */
} dg_status;

typedef enum dg_index_kind {
  DG_INDEX_FLAT = 0,     /* VectorIndexFlat: faiss IndexFlatL2/IP + IDMap2,
                            src/vector/vector_index_flat.cc:81-98 */
  DG_INDEX_IVF_FLAT = 1, /* VectorIndexIvfFlat: faiss IndexIVFFlat,
                            src/vector/vector_index_ivf_flat.cc:805-823 */
  DG_INDEX_IVF_PQ = 2    /* VectorIndexRawIvfPq: faiss IndexIVFPQ,
                            src/vector/vector_index_raw_ivf_pq.cc:551-570 */
} dg_index_kind;

typedef enum dg_metric {
  DG_METRIC_L2 = 0,     /* METRIC_TYPE_L2 */
  DG_METRIC_IP = 1,     /* METRIC_TYPE_INNER_PRODUCT */
  DG_METRIC_COSINE = 2  /* METRIC_TYPE_COSINE = normalize + IP,
                           src/vector/vector_index_flat.cc:88-91 */
} dg_metric;

typedef struct dg_index dg_index; /* opaque */

typedef struct dg_index_desc {
  int32_t kind;      /* dg_index_kind */
  int32_t metric;    /* dg_metric */
  int32_t d;         /* dimension */
  int32_t nlist;     /* IVF: ncentroids (reference default 2048,
                        src/vector/vector_index.h "kCreateIvfFlatParamNcentroids") */
  int32_t pq_m;      /* IVF-PQ: subquantizers (default 64) */
  int32_t pq_nbits;  /* IVF-PQ: bits per code (default 8) */
  int32_t device;    /* HIP device ordinal; -1 = current device */
  int64_t reserve;   /* capacity hint in vectors; 0 = grow on demand */
} dg_index_desc;

/* FilterFunctor forms, reference src/vector/vector_index.h:67-146:
 * RANGE      = RangeFilterFunctor   (min <= id < max), vector_index.h:77-80
 * SORTED_IDS = SortFilterFunctor    (binary search),   vector_index.h:109-146
 * BITMAP     = ConcreteFilterFunctor/IDSelectorBatch as a device bitmap
 * negate mirrors is_negation_ (vector_index.h:88-101). */
typedef enum dg_filter_kind {
  DG_FILTER_NONE = 0,
  DG_FILTER_RANGE = 1,
  DG_FILTER_SORTED_IDS = 2,
  DG_FILTER_BITMAP = 3
} dg_filter_kind;

typedef struct dg_filter {
  int32_t kind;            /* dg_filter_kind */
  int32_t negate;          /* 0 or 1 */
  int64_t min_id, max_id;  /* RANGE: [min_id, max_id) */
  const int64_t* ids;      /* SORTED_IDS: ascending, caller-owned */
  int64_t n_ids;
  const uint64_t* bitmap;  /* BITMAP: bit (id - bitmap_base) */
  int64_t bitmap_base;
  int64_t bitmap_nbits;
} dg_filter;

typedef struct dg_stats_out {
  int64_t ntotal;          /* GetCount, vector_index.h:150 */
  int32_t d;               /* GetDimension, vector_index.h:148 */
  int32_t metric;
  int32_t kind;
  int32_t nlist;
  int32_t is_trained;      /* IsTrained, vector_index.h:201 */
  int64_t device_bytes;    /* GetMemorySize, vector_index.h:152 */
  /* per-stage timings of the LAST dg_search, measured with hipEvents on the
   * index's own stream (GPU equivalent of the Tracker phase recorders,
   * src/common/tracker.h:131-190) */
  double last_coarse_ms;
  double last_scan_ms;     /* dominant kernel: inverted-list scan (IVF) or
                              flat distance+select (Flat) */
  double last_select_ms;
  double last_total_ms;
  int64_t last_nq;
  /* roofline accounting for the dominant kernel of the last search:
   * algorithmic bytes = one read of every probed list's vectors+ids
   * (grouped accounting, SURVEY.md §8d cfg C) */
  int64_t last_scan_bytes_algorithmic;
  double last_scan_gbps_algorithmic;
  /* tombstoned-but-uncompacted rows (GetDeletedCount, vector_index.h:151;
   * the reference's faiss path compacts on remove so it reports 0 — here
   * removes tombstone until the next finalize) */
  int64_t deleted_count;
} dg_stats_out;

/* ---- lifecycle ---- */
dg_status dg_index_create(dg_index** out, const dg_index_desc* desc);
void dg_index_destroy(dg_index* idx);

/* ---- train (IVF): k-means over n x d train vectors.
 * Semantics restated from src/vector/vector_index_ivf_flat.cc:644-712:
 * cosine => normalize train data first (:687-691); n < nlist => nlist
 * degrades to 1 (:676-680); already trained => OK no-op (:669-671).
 * faiss Clustering defaults (niter=25, max 256 points/centroid subsample,
 * seed 1234) restated in oracle/oracle.c and matched here. ---- */
dg_status dg_train(dg_index* idx, int64_t n, const float* x);
/* Inject/extract centroids so CPU oracle and GPU search identical structures
 * (BASELINE.md protocol).  set_centroids marks the index trained. */
dg_status dg_set_centroids(dg_index* idx, int32_t nlist, const float* centroids);
dg_status dg_get_centroids(dg_index* idx, float* out_centroids);
/* IVF-PQ codebook injection/extraction (m x 256 x (d/m) floats), the PQ
 * analog of dg_set_centroids for oracle-shared parity (requires centroids
 * set first; rebuilds the ADC tables). */
dg_status dg_set_codebooks(dg_index* idx, int32_t m, int32_t nbits,
                           const float* codebooks);
dg_status dg_get_codebooks(dg_index* idx, float* out_codebooks);

/* ---- mutation (exclusive) ----
 * add restates faiss add_with_ids via VectorIndexIvfFlat::Add
 * (src/vector/vector_index_ivf_flat.cc:119-124): assign to nearest centroid,
 * append (vector, id) to that inverted list.  Duplicate ids in one call =>
 * DG_EID_DUPLICATED (src/vector/vector_index_utils.cc CheckVectorIdDuplicated).
 * upsert = remove-if-present + add (VectorIndexFlat::Upsert semantics). */
dg_status dg_add(dg_index* idx, int64_t n, const int64_t* ids, const float* x);
/* device-pointer add: d_x lives on the index's device (bench hot-path
 * ingestion; semantics identical to dg_add) */
dg_status dg_add_device(dg_index* idx, int64_t n, const int64_t* ids,
                        const float* d_x);
/* download the per-vector coarse assignment (arrival order, n = ntotal
 * including tombstones).  Lets the CPU-baseline leg rebuild the identical
 * IVF structure without re-running assignment on host. */
dg_status dg_export_assign(dg_index* idx, int32_t* out /* ntotal */);
dg_status dg_upsert(dg_index* idx, int64_t n, const int64_t* ids, const float* x);
/* remove: all ids must exist, else DG_ENOT_FOUND and nothing is removed
 * (src/vector/vector_index_ivf_flat.cc:177-186). */
dg_status dg_remove(dg_index* idx, int64_t n, const int64_t* ids);

/* ---- search ----
 * Restates VectorIndexIvfFlat::Search (src/vector/vector_index_ivf_flat.cc:
 * 191-275): nprobe <= 0 => index default (80, clamped); nprobe clamped to
 * nlist (:234); untrained IVF => all ids -1, DG_OK (:223-227 "direct return
 * blank"); k <= 0 => DG_OK no-op (:201).  Flat ignores nprobe
 * (src/vector/vector_index_flat.cc:205-264).  Cosine queries are normalized
 * like ExtractVectorValue (src/vector/vector_index_utils.cc:564-609,480-491).
 * Host-pointer form: copies in/out and synchronizes. */
dg_status dg_search(dg_index* idx, int64_t nq, const float* x, int32_t k,
                    int32_t nprobe, const dg_filter* filter,
                    float* out_dist /* nq x k */,
                    int64_t* out_ids /* nq x k */);
/* Device-pointer form: x/out_dist/out_ids are device pointers on the index's
 * device; enqueues on the index stream, no synchronization (call dg_sync).
 * This is the resident-in-HBM hot path bench.py times. */
dg_status dg_search_device(dg_index* idx, int64_t nq, const float* d_x,
                           int32_t k, int32_t nprobe, const dg_filter* filter,
                           float* d_out_dist, int64_t* d_out_ids);
dg_status dg_sync(dg_index* idx);

/* ---- range search ----
 * Restates VectorIndexIvfFlat::RangeSearch (vector_index_ivf_flat.cc:278-368):
 * radius in faiss convention (L2: dist < radius, IP/cos: score > radius; the
 * C++ shim applies the 1-r flip of :302-305); results best-first per query.
 * CSR out params: caller frees *out_ids / *out_dists with dg_free. ---- */
dg_status dg_range_search(dg_index* idx, int64_t nq, const float* x,
                          float radius, const dg_filter* filter,
                          int64_t* lims /* nq+1 */, int64_t** out_ids,
                          float** out_dists);
void dg_free(void* p);

/* ---- persistence (Save/Load, vector_index.h:168-170; snapshot files,
 * src/vector/vector_index_snapshot_manager.cc:583-599).  Own container
 * format v1 (documented in DESIGN.md) plus the faiss-compatible container
 * below. */
dg_status dg_save(dg_index* idx, const char* path);
dg_status dg_load(dg_index** out, const char* path, int32_t device);

/* ---- faiss-file-compatible persistence ----
 * The reference's snapshot/install cycle ships faiss::write_index files
 * (vector_index_snapshot_manager.cc:583-599; written at
 * vector_index_flat.cc:354, vector_index_ivf_flat.cc:398, loaded at
 * vector_index_flat.cc:368-462, vector_index_ivf_flat.cc:413-514).  These
 * functions emit/ingest the faiss 1.7.x container byte layout (the
 * reference's faiss fork is >= 1.7.3 — it uses faiss::SearchParameters,
 * vector_index_flat.cc:232):
 *   FLAT     -> IndexIDMap2{IndexFlatL2|IndexFlatIP}   ("IxM2"/"IxF2"/"IxFI")
 *   IVF_FLAT -> IndexIVFFlat                            ("IwFl")
 *   IVF_PQ   -> IndexIVFPQ (residual, nbits=8)          ("IwPQ")
 * Cosine indexes are written as IP over stored (normalized) vectors, like
 * the reference (vector_index_flat.cc:88-91); on load, pass
 * metric_override = DG_METRIC_COSINE to reconstruct a cosine index from an
 * IP-metric file (-1 keeps the file metric).  Tombstoned rows are dropped
 * at save.  Format notes and the byte-level layout are in DESIGN.md. */
dg_status dg_save_faiss(dg_index* idx, const char* path);
dg_status dg_load_faiss(dg_index** out, const char* path,
                        int32_t metric_override, int32_t device);

/* ---- multi-GPU sharding support ----
 * Each rank holds a shard of the database (row-sharded)
 * with replicated centroids; local top-k is merged with an RCCL all-gather by
 * the caller.  An optional list mask restricts scanning to owned lists for
 * list-sharded deployments: mask[l] != 0 => list l is scanned here. */
dg_status dg_set_list_mask(dg_index* idx, const uint8_t* mask /* nlist */);

/* ---- wrapper-lifecycle lock (LockWrite/UnlockWrite,
 * vector_index.h:192-193): the exclusive lock the reference wrapper takes
 * around its fork-save window; must be released by the same thread. ---- */
void dg_lock_write(dg_index* idx);
void dg_unlock_write(dg_index* idx);

/* ---- introspection ---- */
dg_status dg_stats(dg_index* idx, dg_stats_out* out);
void dg_last_error(char* buf, int64_t len);
int dg_device_count(void);
/* Library self-identification: returns a static string with build arch. */
const char* dg_build_info(void);

#ifdef __cplusplus
} /* extern "C" */
#endif
#endif /* DINGO_GPU_H_ */
