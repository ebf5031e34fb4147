/* oracle.c — CPU restatement of the dingo-store vector-search hot path.
 * See oracle.h for scope, pinning status and the only-for-tests rule.
 *
 * Build: strict FP (no -ffast-math) so the arithmetic core stays bit-exact
 * vs /root/reference/src/simd/distances_ref.cc compiled into oracle/_ref.
 * The faster reordered-sum search used only as bench.py's cpu_baseline leg
 * lives in oracle_fast.c.
 */
#include "oracle.h"

#include <math.h>
#include <stdlib.h>
#include <string.h>

#ifdef _OPENMP
#include <omp.h>
#endif

const char* dgo_version(void) { return "dgo-oracle-0.1 strict"; }

/* ---------------- arithmetic core ----------------
 * Bit-exact restatement of src/simd/distances_ref.cc (same loop order,
 * same accumulator types — note the double accumulator in norm_L2sqr,
 * distances_ref.cc:62-66). */
float dgo_fvec_L2sqr(const float* x, const float* y, size_t d) {
  /* distances_ref.cc:24-31 */
  size_t i;
  float res = 0;
  for (i = 0; i < d; i++) {
    const float tmp = x[i] - y[i];
    res += tmp * tmp;
  }
  return res;
}

float dgo_fvec_inner_product(const float* x, const float* y, size_t d) {
  /* distances_ref.cc:53-58 */
  size_t i;
  float res = 0;
  for (i = 0; i < d; i++) res += x[i] * y[i];
  return res;
}

float dgo_fvec_norm_L2sqr(const float* x, size_t d) {
  /* distances_ref.cc:60-66 — double accumulator, float return */
  size_t i;
  double res = 0;
  for (i = 0; i < d; i++) res += x[i] * x[i];
  return res;
}

void dgo_normalize(float* x, int32_t d) {
  /* VectorIndexUtils::NormalizeVectorForFaiss,
   * src/vector/vector_index_utils.cc:480-491: skip when already unit norm
   * within 1e-5, divide by sqrt otherwise. */
  static const float kFloatAccuracy = 0.00001f;
  float norm_l2_sqr = dgo_fvec_norm_L2sqr(x, d);
  if (norm_l2_sqr > 0 && fabsf(1.0f - norm_l2_sqr) > kFloatAccuracy) {
    float norm_l2 = sqrtf(norm_l2_sqr);
    for (int32_t i = 0; i < d; i++) x[i] = x[i] / norm_l2;
  }
}

void dgo_normalize_batch(float* x, int64_t n, int32_t d) {
#pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < n; i++) dgo_normalize(x + (size_t)i * d, d);
}

/* ---------------- top-k helper ----------------
 * Smaller key = better.  L2: key = dist.  IP/cosine: key = -score.
 * Ties broken toward smaller id (deterministic; parity comparisons allow
 * fp-tie reordering). */
typedef struct {
  float key;
  int64_t id;
} dgo_cand;

typedef struct {
  dgo_cand* c;
  int32_t k, size, worst; /* worst = index of current worst entry */
} dgo_topk;

static inline int cand_worse(const dgo_cand* a, const dgo_cand* b) {
  /* is a worse than b? */
  if (a->key != b->key) return a->key > b->key;
  return a->id > b->id;
}

static inline void topk_init(dgo_topk* t, dgo_cand* storage, int32_t k) {
  t->c = storage;
  t->k = k;
  t->size = 0;
  t->worst = 0;
}

static inline void topk_recompute_worst(dgo_topk* t) {
  int w = 0;
  for (int i = 1; i < t->size; i++)
    if (cand_worse(&t->c[i], &t->c[w])) w = i;
  t->worst = w;
}

static inline void topk_push(dgo_topk* t, float key, int64_t id) {
  dgo_cand nc = {key, id};
  if (t->size < t->k) {
    t->c[t->size] = nc;
    if (t->size == 0 || cand_worse(&nc, &t->c[t->worst])) t->worst = t->size;
    t->size++;
  } else if (cand_worse(&t->c[t->worst], &nc)) {
    t->c[t->worst] = nc;
    topk_recompute_worst(t);
  }
}

static int cand_cmp(const void* pa, const void* pb) {
  const dgo_cand* a = (const dgo_cand*)pa;
  const dgo_cand* b = (const dgo_cand*)pb;
  if (a->key < b->key) return -1;
  if (a->key > b->key) return 1;
  if (a->id < b->id) return -1;
  if (a->id > b->id) return 1;
  return 0;
}

/* finalize: ascending by (key,id); emit faiss-convention distance and id,
 * pad with -1 / 0 like the reference label prefill
 * (vector_index_flat.cc:218). */
static void topk_emit(dgo_topk* t, int metric, float* out_dist,
                      int64_t* out_ids) {
  qsort(t->c, t->size, sizeof(dgo_cand), cand_cmp);
  for (int i = 0; i < t->k; i++) {
    if (i < t->size) {
      out_dist[i] = (metric == DGO_L2) ? t->c[i].key : -t->c[i].key;
      out_ids[i] = t->c[i].id;
    } else {
      out_dist[i] = 0.0f;
      out_ids[i] = -1;
    }
  }
}

static inline float metric_key(int metric, const float* q, const float* v,
                               int32_t d) {
  if (metric == DGO_L2) return dgo_fvec_L2sqr(q, v, d);
  return -dgo_fvec_inner_product(q, v, d); /* IP and COSINE */
}

/* ---------------- Flat search ---------------- */
void dgo_flat_search(int metric, int64_t n, int32_t d, const float* base,
                     const int64_t* ids, int64_t nq, const float* queries,
                     int32_t k, float* out_dist, int64_t* out_ids) {
#pragma omp parallel
  {
    dgo_cand* storage = (dgo_cand*)malloc(sizeof(dgo_cand) * k);
#pragma omp for schedule(dynamic, 1)
    for (int64_t q = 0; q < nq; q++) {
      const float* qv = queries + (size_t)q * d;
      dgo_topk t;
      topk_init(&t, storage, k);
      for (int64_t i = 0; i < n; i++) {
        float key = metric_key(metric, qv, base + (size_t)i * d, d);
        topk_push(&t, key, ids ? ids[i] : i);
      }
      topk_emit(&t, metric, out_dist + (size_t)q * k, out_ids + (size_t)q * k);
    }
    free(storage);
  }
}

/* ---------------- mt19937 + faiss-style rand helpers ----------------
 * Standard MT19937 (= std::mt19937 as used by faiss RandomGenerator). */
typedef struct {
  uint32_t mt[624];
  int idx;
} dgo_mt;

static void mt_seed(dgo_mt* s, uint32_t seed) {
  s->mt[0] = seed;
  for (int i = 1; i < 624; i++)
    s->mt[i] = 1812433253u * (s->mt[i - 1] ^ (s->mt[i - 1] >> 30)) + i;
  s->idx = 624;
}

static uint32_t mt_next(dgo_mt* s) {
  if (s->idx >= 624) {
    for (int i = 0; i < 624; i++) {
      uint32_t y = (s->mt[i] & 0x80000000u) | (s->mt[(i + 1) % 624] & 0x7fffffffu);
      s->mt[i] = s->mt[(i + 397) % 624] ^ (y >> 1);
      if (y & 1) s->mt[i] ^= 2567483615u;
    }
    s->idx = 0;
  }
  uint32_t y = s->mt[s->idx++];
  y ^= y >> 11;
  y ^= (y << 7) & 2636928640u;
  y ^= (y << 15) & 4022730752u;
  y ^= y >> 18;
  return y;
}

/* faiss RandomGenerator::rand_int(max) = mt() % max; rand_float = mt()/2^32 */
static inline int64_t mt_rand_int(dgo_mt* s, int64_t max) {
  return (int64_t)(mt_next(s) % (uint64_t)max);
}
static inline float mt_rand_float(dgo_mt* s) {
  return mt_next(s) * (1.0f / 4294967296.0f);
}

/* faiss rand_perm: identity then Fisher-Yates with rand_int(n-i) */
static void rand_perm(int64_t* perm, int64_t n, uint32_t seed) {
  for (int64_t i = 0; i < n; i++) perm[i] = i;
  dgo_mt rng;
  mt_seed(&rng, seed);
  for (int64_t i = 0; i + 1 < n; i++) {
    int64_t i2 = i + mt_rand_int(&rng, n - i);
    int64_t tmp = perm[i];
    perm[i] = perm[i2];
    perm[i2] = tmp;
  }
}

/* ---------------- coarse assignment ---------------- */
static inline int32_t nearest_centroid(int metric, const float* v,
                                       const float* centroids, int32_t nlist,
                                       int32_t d) {
  int32_t best = 0;
  float bkey = metric_key(metric, v, centroids, d);
  for (int32_t l = 1; l < nlist; l++) {
    float key = metric_key(metric, v, centroids + (size_t)l * d, d);
    if (key < bkey) { /* strict <: first best wins, like faiss heap replace */
      bkey = key;
      best = l;
    }
  }
  return best;
}

void dgo_ivf_assign(int metric, int64_t n, int32_t d, const float* x,
                    int32_t nlist, const float* centroids,
                    int32_t* assign_out) {
#pragma omp parallel for schedule(dynamic, 256)
  for (int64_t i = 0; i < n; i++)
    assign_out[i] = nearest_centroid(metric, x + (size_t)i * d, centroids,
                                     nlist, d);
}

/* ---------------- k-means (faiss Clustering restated) ----------------
 * Defaults restated from faiss ClusteringParameters (referenced at
 * src/vector/vector_index_ivf_flat.cc:654-664): niter=25, seed=1234,
 * max_points_per_centroid=256 (subsample above that), empty clusters split
 * from a probabilistically chosen large cluster with ±1/1024 perturbation.
 * Deterministic given (seed, metric, data). */
void dgo_kmeans(int metric, int64_t n, int32_t d, const float* x,
                int32_t nlist, int32_t niter, uint32_t seed,
                float* centroids_out) {
  const int64_t max_points = (int64_t)256 * nlist;
  const float EPS = 1.0f / 1024.0f;

  /* subsample_training_set: perm with seed, take first max_points */
  const float* data = x;
  float* subsampled = NULL;
  int64_t nt = n;
  if (n > max_points) {
    int64_t* perm = (int64_t*)malloc(sizeof(int64_t) * n);
    rand_perm(perm, n, seed);
    subsampled = (float*)malloc(sizeof(float) * (size_t)max_points * d);
    for (int64_t i = 0; i < max_points; i++)
      memcpy(subsampled + (size_t)i * d, x + (size_t)perm[i] * d,
             sizeof(float) * d);
    free(perm);
    data = subsampled;
    nt = max_points;
  }

  /* init centroids: first nlist of a fresh permutation (faiss uses
   * seed + 1 + redo * 15486557L with redo=0) */
  {
    int64_t* perm = (int64_t*)malloc(sizeof(int64_t) * nt);
    rand_perm(perm, nt, seed + 1);
    for (int32_t c = 0; c < nlist; c++)
      memcpy(centroids_out + (size_t)c * d, data + (size_t)perm[c] * d,
             sizeof(float) * d);
    free(perm);
  }

  int32_t* assign = (int32_t*)malloc(sizeof(int32_t) * nt);
  double* sums = (double*)malloc(sizeof(double) * (size_t)nlist * d);
  int64_t* hist = (int64_t*)malloc(sizeof(int64_t) * nlist);
  dgo_mt split_rng;
  mt_seed(&split_rng, seed + 2);

  for (int32_t iter = 0; iter < niter; iter++) {
    dgo_ivf_assign(metric, nt, d, data, nlist, centroids_out, assign);

    memset(sums, 0, sizeof(double) * (size_t)nlist * d);
    memset(hist, 0, sizeof(int64_t) * nlist);
    for (int64_t i = 0; i < nt; i++) {
      int32_t c = assign[i];
      hist[c]++;
      const float* v = data + (size_t)i * d;
      double* s = sums + (size_t)c * d;
      for (int32_t j = 0; j < d; j++) s[j] += v[j];
    }
    for (int32_t c = 0; c < nlist; c++) {
      if (hist[c] > 0) {
        float* cent = centroids_out + (size_t)c * d;
        const double* s = sums + (size_t)c * d;
        for (int32_t j = 0; j < d; j++) cent[j] = (float)(s[j] / hist[c]);
      }
    }
    /* split_clusters (faiss Clustering.cpp semantics) */
    for (int32_t ci = 0; ci < nlist; ci++) {
      if (hist[ci] != 0) continue;
      int32_t cj = 0;
      for (;; cj = (cj + 1) % nlist) {
        float p = (hist[cj] - 1.0f) / (float)(nt - nlist);
        if (mt_rand_float(&split_rng) < p) break;
      }
      memcpy(centroids_out + (size_t)ci * d, centroids_out + (size_t)cj * d,
             sizeof(float) * d);
      for (int32_t j = 0; j < d; j++) {
        if (j % 2 == 0) {
          centroids_out[(size_t)ci * d + j] *= 1 + EPS;
          centroids_out[(size_t)cj * d + j] *= 1 - EPS;
        } else {
          centroids_out[(size_t)ci * d + j] *= 1 - EPS;
          centroids_out[(size_t)cj * d + j] *= 1 + EPS;
        }
      }
      hist[ci] = hist[cj] / 2;
      hist[cj] -= hist[ci];
    }
  }
  free(assign);
  free(sums);
  free(hist);
  free(subsampled);
}

/* ---------------- CSR build ---------------- */
void dgo_ivf_build(int64_t n, int32_t d, const float* x, const int64_t* ids,
                   int32_t nlist, const int32_t* assign, int64_t* offsets_out,
                   float* grouped_vectors_out, int64_t* grouped_ids_out) {
  memset(offsets_out, 0, sizeof(int64_t) * (nlist + 1));
  for (int64_t i = 0; i < n; i++) offsets_out[assign[i] + 1]++;
  for (int32_t l = 0; l < nlist; l++) offsets_out[l + 1] += offsets_out[l];
  int64_t* cursor = (int64_t*)malloc(sizeof(int64_t) * nlist);
  memcpy(cursor, offsets_out, sizeof(int64_t) * nlist);
  for (int64_t i = 0; i < n; i++) {
    int64_t pos = cursor[assign[i]]++;
    memcpy(grouped_vectors_out + (size_t)pos * d, x + (size_t)i * d,
           sizeof(float) * d);
    grouped_ids_out[pos] = ids ? ids[i] : i;
  }
  free(cursor);
}

/* ---------------- coarse probe ---------------- */
void dgo_coarse_probe(int metric, int32_t nlist, int32_t d,
                      const float* centroids, int64_t nq,
                      const float* queries, int32_t nprobe,
                      int32_t* probes_out) {
#pragma omp parallel
  {
    dgo_cand* storage = (dgo_cand*)malloc(sizeof(dgo_cand) * nprobe);
#pragma omp for schedule(dynamic, 8)
    for (int64_t q = 0; q < nq; q++) {
      const float* qv = queries + (size_t)q * d;
      dgo_topk t;
      topk_init(&t, storage, nprobe);
      for (int32_t l = 0; l < nlist; l++)
        topk_push(&t, metric_key(metric, qv, centroids + (size_t)l * d, d), l);
      qsort(t.c, t.size, sizeof(dgo_cand), cand_cmp);
      for (int32_t p = 0; p < nprobe; p++)
        probes_out[q * nprobe + p] = (p < t.size) ? (int32_t)t.c[p].id : -1;
    }
    free(storage);
  }
}

/* ---------------- IVF-Flat search ---------------- */
void dgo_ivf_search(int metric, int32_t nlist, int32_t d,
                    const float* centroids, const int64_t* offsets,
                    const float* grouped_vectors, const int64_t* grouped_ids,
                    int64_t nq, const float* queries, int32_t k,
                    int32_t nprobe, const uint8_t* list_mask,
                    float* out_dist, int64_t* out_ids) {
  if (nprobe > nlist) nprobe = nlist; /* clamp, ivf_flat.cc:234 */
#pragma omp parallel
  {
    dgo_cand* pstorage = (dgo_cand*)malloc(sizeof(dgo_cand) * nprobe);
    dgo_cand* kstorage = (dgo_cand*)malloc(sizeof(dgo_cand) * k);
#pragma omp for schedule(dynamic, 1)
    for (int64_t q = 0; q < nq; q++) {
      const float* qv = queries + (size_t)q * d;
      dgo_topk probes;
      topk_init(&probes, pstorage, nprobe);
      for (int32_t l = 0; l < nlist; l++)
        topk_push(&probes, metric_key(metric, qv, centroids + (size_t)l * d, d),
                  l);
      qsort(probes.c, probes.size, sizeof(dgo_cand), cand_cmp);

      dgo_topk t;
      topk_init(&t, kstorage, k);
      for (int32_t p = 0; p < probes.size; p++) {
        int32_t l = (int32_t)probes.c[p].id;
        if (list_mask && !list_mask[l]) continue;
        for (int64_t i = offsets[l]; i < offsets[l + 1]; i++) {
          float key =
              metric_key(metric, qv, grouped_vectors + (size_t)i * d, d);
          topk_push(&t, key, grouped_ids[i]);
        }
      }
      topk_emit(&t, metric, out_dist + (size_t)q * k, out_ids + (size_t)q * k);
    }
    free(pstorage);
    free(kstorage);
  }
}

/* ---------------- IVF-PQ ----------------
 * faiss IndexIVFPQ restated (by_residual=true default): codes are
 * per-subspace nearest codebook entries of (x - coarse_centroid); search is
 * ADC with a per-(query,list) LUT.  Reference call sites:
 * src/vector/vector_index_raw_ivf_pq.cc:157-210 (search), :476 (train). */
void dgo_pq_train(int64_t n, int32_t d, const float* residuals, int32_t m,
                  int32_t nbits, uint32_t seed, float* codebooks_out) {
  const int32_t ksub = 1 << nbits;
  const int32_t dsub = d / m;
  /* faiss ProductQuantizer::train: one k-means per subspace over the
   * subvectors (niter 25, same Clustering defaults). */
  float* sub = (float*)malloc(sizeof(float) * (size_t)n * dsub);
  for (int32_t mi = 0; mi < m; mi++) {
    for (int64_t i = 0; i < n; i++)
      memcpy(sub + (size_t)i * dsub, residuals + (size_t)i * d + mi * dsub,
             sizeof(float) * dsub);
    dgo_kmeans(DGO_L2, n, dsub, sub, ksub, 25, seed,
               codebooks_out + (size_t)mi * ksub * dsub);
  }
  free(sub);
}

void dgo_ivfpq_encode(int64_t n, int32_t d, const float* x,
                      const int32_t* assign, const float* centroids,
                      int32_t m, const float* codebooks, uint8_t* codes_out) {
  const int32_t ksub = 256;
  const int32_t dsub = d / m;
#pragma omp parallel
  {
    float* res = (float*)malloc(sizeof(float) * d);
#pragma omp for schedule(dynamic, 64)
    for (int64_t i = 0; i < n; i++) {
      const float* v = x + (size_t)i * d;
      const float* c = centroids + (size_t)assign[i] * d;
      for (int32_t j = 0; j < d; j++) res[j] = v[j] - c[j];
      for (int32_t mi = 0; mi < m; mi++) {
        const float* cb = codebooks + (size_t)mi * ksub * dsub;
        int32_t best = 0;
        float bkey = dgo_fvec_L2sqr(res + mi * dsub, cb, dsub);
        for (int32_t kk = 1; kk < ksub; kk++) {
          float key = dgo_fvec_L2sqr(res + mi * dsub, cb + (size_t)kk * dsub,
                                     dsub);
          if (key < bkey) {
            bkey = key;
            best = kk;
          }
        }
        codes_out[(size_t)i * m + mi] = (uint8_t)best;
      }
    }
    free(res);
  }
}

void dgo_ivfpq_search(int metric, int32_t nlist, int32_t d,
                      const float* centroids, const int64_t* offsets,
                      const uint8_t* grouped_codes, const int64_t* grouped_ids,
                      int32_t m, const float* codebooks, int64_t nq,
                      const float* queries, int32_t k, int32_t nprobe,
                      float* out_dist, int64_t* out_ids) {
  const int32_t ksub = 256;
  const int32_t dsub = d / m;
  if (nprobe > nlist) nprobe = nlist;
#pragma omp parallel
  {
    dgo_cand* pstorage = (dgo_cand*)malloc(sizeof(dgo_cand) * nprobe);
    dgo_cand* kstorage = (dgo_cand*)malloc(sizeof(dgo_cand) * k);
    float* lut = (float*)malloc(sizeof(float) * (size_t)m * ksub);
    float* res = (float*)malloc(sizeof(float) * d);
#pragma omp for schedule(dynamic, 1)
    for (int64_t q = 0; q < nq; q++) {
      const float* qv = queries + (size_t)q * d;
      dgo_topk probes;
      topk_init(&probes, pstorage, nprobe);
      for (int32_t l = 0; l < nlist; l++)
        topk_push(&probes, metric_key(metric, qv, centroids + (size_t)l * d, d),
                  l);
      qsort(probes.c, probes.size, sizeof(dgo_cand), cand_cmp);

      dgo_topk t;
      topk_init(&t, kstorage, k);
      for (int32_t p = 0; p < probes.size; p++) {
        int32_t l = (int32_t)probes.c[p].id;
        const float* cl = centroids + (size_t)l * d;
        float bias = 0.0f;
        if (metric == DGO_L2) {
          /* LUT[mi][kk] = || (q - c_l)_mi - cb ||^2 */
          for (int32_t j = 0; j < d; j++) res[j] = qv[j] - cl[j];
          for (int32_t mi = 0; mi < m; mi++) {
            const float* cb = codebooks + (size_t)mi * ksub * dsub;
            for (int32_t kk = 0; kk < ksub; kk++)
              lut[mi * ksub + kk] =
                  dgo_fvec_L2sqr(res + mi * dsub, cb + (size_t)kk * dsub, dsub);
          }
        } else {
          /* score = q.c_l + sum_mi q_mi . cb  (residual IP decomposition) */
          bias = dgo_fvec_inner_product(qv, cl, d);
          for (int32_t mi = 0; mi < m; mi++) {
            const float* cb = codebooks + (size_t)mi * ksub * dsub;
            for (int32_t kk = 0; kk < ksub; kk++)
              lut[mi * ksub + kk] = dgo_fvec_inner_product(
                  qv + mi * dsub, cb + (size_t)kk * dsub, dsub);
          }
        }
        for (int64_t i = offsets[l]; i < offsets[l + 1]; i++) {
          const uint8_t* code = grouped_codes + (size_t)i * m;
          float acc = 0.0f;
          for (int32_t mi = 0; mi < m; mi++) acc += lut[mi * ksub + code[mi]];
          float key = (metric == DGO_L2) ? acc : -(bias + acc);
          topk_push(&t, key, grouped_ids[i]);
        }
      }
      topk_emit(&t, metric, out_dist + (size_t)q * k, out_ids + (size_t)q * k);
    }
    free(pstorage);
    free(kstorage);
    free(lut);
    free(res);
  }
}

/* ---------------- range search ----------------
 * faiss RangeSearch semantics restated (reference RangeSearch call sites,
 * vector_index_flat.cc:268+ / vector_index_ivf_flat.cc:278-368): L2 keeps
 * dist < radius, IP/cos keeps score > radius (the 1-r flip happens in the
 * shim).  Results per query sorted best-first, ties toward smaller id.
 * Caller provides capacity; results beyond cap are counted in lims but
 * dropped (tests size cap generously). */
static void range_collect(int metric, const float* qv, int32_t d,
                          const float* vecs, const int64_t* ids_arr,
                          int64_t lo, int64_t hi, float radius,
                          dgo_cand** buf, int64_t* cnt, int64_t* cap_now) {
  float thr = (metric == DGO_L2) ? radius : -radius;
  for (int64_t i = lo; i < hi; i++) {
    float key = metric_key(metric, qv, vecs + (size_t)i * d, d);
    if (key < thr) {
      if (*cnt == *cap_now) {
        *cap_now = *cap_now * 2 + 64;
        *buf = (dgo_cand*)realloc(*buf, sizeof(dgo_cand) * *cap_now);
      }
      (*buf)[*cnt].key = key;
      (*buf)[*cnt].id = ids_arr ? ids_arr[i] : i;
      (*cnt)++;
    }
  }
}

void dgo_flat_range_search(int metric, int64_t n, int32_t d,
                           const float* base, const int64_t* ids, int64_t nq,
                           const float* queries, float radius, int64_t* lims,
                           int64_t cap, float* out_dist, int64_t* out_ids) {
  lims[0] = 0;
  for (int64_t q = 0; q < nq; q++) {
    dgo_cand* buf = NULL;
    int64_t cnt = 0, cap_now = 0;
    range_collect(metric, queries + (size_t)q * d, d, base, ids, 0, n,
                  radius, &buf, &cnt, &cap_now);
    qsort(buf, cnt, sizeof(dgo_cand), cand_cmp);
    int64_t base_off = lims[q];
    for (int64_t i = 0; i < cnt && base_off + i < cap; i++) {
      out_dist[base_off + i] =
          (metric == DGO_L2) ? buf[i].key : -buf[i].key;
      out_ids[base_off + i] = buf[i].id;
    }
    lims[q + 1] = base_off + cnt;
    free(buf);
  }
}

void dgo_ivf_range_search(int metric, int32_t nlist, int32_t d,
                          const float* centroids, const int64_t* offsets,
                          const float* grouped_vectors,
                          const int64_t* grouped_ids, int64_t nq,
                          const float* queries, float radius, int32_t nprobe,
                          int64_t* lims, int64_t cap, float* out_dist,
                          int64_t* out_ids) {
  if (nprobe > nlist) nprobe = nlist;
  lims[0] = 0;
  dgo_cand* pstorage = (dgo_cand*)malloc(sizeof(dgo_cand) * nprobe);
  for (int64_t q = 0; q < nq; q++) {
    const float* qv = queries + (size_t)q * d;
    dgo_topk probes;
    topk_init(&probes, pstorage, nprobe);
    for (int32_t l = 0; l < nlist; l++)
      topk_push(&probes, metric_key(metric, qv, centroids + (size_t)l * d, d),
                l);
    qsort(probes.c, probes.size, sizeof(dgo_cand), cand_cmp);
    dgo_cand* buf = NULL;
    int64_t cnt = 0, cap_now = 0;
    for (int32_t p = 0; p < probes.size; p++) {
      int32_t l = (int32_t)probes.c[p].id;
      range_collect(metric, qv, d, grouped_vectors, grouped_ids, offsets[l],
                    offsets[l + 1], radius, &buf, &cnt, &cap_now);
    }
    qsort(buf, cnt, sizeof(dgo_cand), cand_cmp);
    int64_t base_off = lims[q];
    for (int64_t i = 0; i < cnt && base_off + i < cap; i++) {
      out_dist[base_off + i] =
          (metric == DGO_L2) ? buf[i].key : -buf[i].key;
      out_ids[base_off + i] = buf[i].id;
    }
    lims[q + 1] = base_off + cnt;
    free(buf);
  }
  free(pstorage);
}
