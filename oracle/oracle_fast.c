/* oracle_fast.c — vectorized (reordered-sum) variants of the oracle search
 * paths, used ONLY as bench.py's cpu_baseline timing leg (the
 * reference's hot loops run AVX512 via src/simd hooks,
 * src/vector/vector_index.cc:148-185, so a scalar strict-FP port would
 * understate the CPU baseline).  Arithmetically the same computation with
 * relaxed summation order; parity tests use the strict oracle.c path.
 * Compiled with -O3 -march=native -ffast-math (Makefile).
 */
#include "oracle.h"

#include <math.h>
#include <stdlib.h>
#include <string.h>

typedef struct {
  float key;
  int64_t id;
} fcand;

static inline float l2sqr_fast(const float* restrict x,
                               const float* restrict y, int32_t d) {
  float res = 0;
  for (int32_t i = 0; i < d; i++) {
    float t = x[i] - y[i];
    res += t * t;
  }
  return res;
}

static inline float ip_fast(const float* restrict x, const float* restrict y,
                            int32_t d) {
  float res = 0;
  for (int32_t i = 0; i < d; i++) res += x[i] * y[i];
  return res;
}

static inline float key_fast(int metric, const float* q, const float* v,
                             int32_t d) {
  return metric == DGO_L2 ? l2sqr_fast(q, v, d) : -ip_fast(q, v, d);
}

static int fcand_cmp(const void* pa, const void* pb) {
  const fcand* a = (const fcand*)pa;
  const fcand* b = (const fcand*)pb;
  if (a->key < b->key) return -1;
  if (a->key > b->key) return 1;
  return (a->id < b->id) ? -1 : (a->id > b->id) ? 1 : 0;
}

static void topk_scan_emit(int metric, const float* qv, int32_t d,
                           const float* vecs, const int64_t* ids,
                           const int64_t* idx_ranges, int32_t n_ranges,
                           int32_t k, float* out_dist, int64_t* out_ids,
                           fcand* heap) {
  int32_t size = 0, worst = 0;
  for (int32_t r = 0; r < n_ranges; r++) {
    for (int64_t i = idx_ranges[2 * r]; i < idx_ranges[2 * r + 1]; i++) {
      float key = key_fast(metric, qv, vecs + (size_t)i * d, d);
      int64_t id = ids ? ids[i] : i;
      if (size < k) {
        heap[size].key = key;
        heap[size].id = id;
        if (size == 0 || key > heap[worst].key ||
            (key == heap[worst].key && id > heap[worst].id))
          worst = size;
        size++;
      } else if (key < heap[worst].key ||
                 (key == heap[worst].key && id < heap[worst].id)) {
        heap[worst].key = key;
        heap[worst].id = id;
        worst = 0;
        for (int32_t j = 1; j < size; j++)
          if (heap[j].key > heap[worst].key ||
              (heap[j].key == heap[worst].key && heap[j].id > heap[worst].id))
            worst = j;
      }
    }
  }
  qsort(heap, size, sizeof(fcand), fcand_cmp);
  for (int32_t i = 0; i < k; i++) {
    if (i < size) {
      out_dist[i] = (metric == DGO_L2) ? heap[i].key : -heap[i].key;
      out_ids[i] = heap[i].id;
    } else {
      out_dist[i] = 0.0f;
      out_ids[i] = -1;
    }
  }
}

void dgo_flat_search_fast(int metric, int64_t n, int32_t d, const float* base,
                          const int64_t* ids, int64_t nq, const float* queries,
                          int32_t k, float* out_dist, int64_t* out_ids) {
#pragma omp parallel
  {
    fcand* heap = (fcand*)malloc(sizeof(fcand) * k);
    int64_t range[2] = {0, n};
#pragma omp for schedule(dynamic, 1)
    for (int64_t q = 0; q < nq; q++)
      topk_scan_emit(metric, queries + (size_t)q * d, d, base, ids, range, 1,
                     k, out_dist + (size_t)q * k, out_ids + (size_t)q * k,
                     heap);
    free(heap);
  }
}

void dgo_ivf_search_indexed_fast(int metric, int32_t nlist, int32_t d,
                                 const float* centroids,
                                 const int64_t* member_offsets,
                                 const int64_t* member_rows,
                                 const float* base /* ungrouped */,
                                 int64_t nq, const float* queries, int32_t k,
                                 int32_t nprobe, float* out_dist,
                                 int64_t* out_ids) {
  /* CPU-baseline variant that scans lists through a row-index indirection
   * (member_rows grouped by list via member_offsets) over the UNGROUPED
   * base array — same arithmetic, avoids a second 30 GB host copy at
   * BASELINE cfg C (BASELINE.md / DESIGN.md §cpu-baseline). ids are the
   * row indices themselves. */
  if (nprobe > nlist) nprobe = nlist;
#pragma omp parallel
  {
    fcand* probes = (fcand*)malloc(sizeof(fcand) * nlist);
    fcand* heap = (fcand*)malloc(sizeof(fcand) * k);
#pragma omp for schedule(dynamic, 1)
    for (int64_t q = 0; q < nq; q++) {
      const float* qv = queries + (size_t)q * d;
      for (int32_t l = 0; l < nlist; l++) {
        probes[l].key = key_fast(metric, qv, centroids + (size_t)l * d, d);
        probes[l].id = l;
      }
      qsort(probes, nlist, sizeof(fcand), fcand_cmp);
      int32_t size = 0, worst = 0;
      for (int32_t p = 0; p < nprobe; p++) {
        int32_t l = (int32_t)probes[p].id;
        for (int64_t m = member_offsets[l]; m < member_offsets[l + 1]; m++) {
          int64_t row = member_rows[m];
          float key = key_fast(metric, qv, base + (size_t)row * d, d);
          if (size < k) {
            heap[size].key = key;
            heap[size].id = row;
            if (size == 0 || key > heap[worst].key) worst = size;
            size++;
          } else if (key < heap[worst].key) {
            heap[worst].key = key;
            heap[worst].id = row;
            worst = 0;
            for (int32_t j = 1; j < size; j++)
              if (heap[j].key > heap[worst].key) worst = j;
          }
        }
      }
      qsort(heap, size, sizeof(fcand), fcand_cmp);
      for (int32_t i = 0; i < k; i++) {
        if (i < size) {
          out_dist[(size_t)q * k + i] =
              (metric == DGO_L2) ? heap[i].key : -heap[i].key;
          out_ids[(size_t)q * k + i] = heap[i].id;
        } else {
          out_dist[(size_t)q * k + i] = 0.0f;
          out_ids[(size_t)q * k + i] = -1;
        }
      }
    }
    free(probes);
    free(heap);
  }
}

void dgo_ivf_search_fast(int metric, int32_t nlist, int32_t d,
                         const float* centroids, const int64_t* offsets,
                         const float* grouped_vectors,
                         const int64_t* grouped_ids, int64_t nq,
                         const float* queries, int32_t k, int32_t nprobe,
                         const uint8_t* list_mask, float* out_dist,
                         int64_t* out_ids) {
  if (nprobe > nlist) nprobe = nlist;
#pragma omp parallel
  {
    fcand* probes = (fcand*)malloc(sizeof(fcand) * nlist);
    fcand* heap = (fcand*)malloc(sizeof(fcand) * k);
    int64_t* ranges = (int64_t*)malloc(sizeof(int64_t) * 2 * nprobe);
#pragma omp for schedule(dynamic, 1)
    for (int64_t q = 0; q < nq; q++) {
      const float* qv = queries + (size_t)q * d;
      for (int32_t l = 0; l < nlist; l++) {
        probes[l].key = key_fast(metric, qv, centroids + (size_t)l * d, d);
        probes[l].id = l;
      }
      qsort(probes, nlist, sizeof(fcand), fcand_cmp);
      int32_t nr = 0;
      for (int32_t p = 0; p < nprobe; p++) {
        int32_t l = (int32_t)probes[p].id;
        if (list_mask && !list_mask[l]) continue;
        ranges[2 * nr] = offsets[l];
        ranges[2 * nr + 1] = offsets[l + 1];
        nr++;
      }
      topk_scan_emit(metric, qv, d, grouped_vectors, grouped_ids, ranges, nr,
                     k, out_dist + (size_t)q * k, out_ids + (size_t)q * k,
                     heap);
    }
    free(probes);
    free(heap);
    free(ranges);
  }
}
