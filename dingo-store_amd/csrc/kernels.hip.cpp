// kernels.hip.cpp — CDNA4 (gfx950) kernels for the dingo-store vector-search
// hot path.  Net-new MI355X-native code (the reference is CPU-only; the
// faiss math it delegates to is restated from SURVEY.md §8a semantics).
//
// Design:
//  * The hot path is HBM-bound (SURVEY.md §8d cfg C): every kernel reads the
//    database stream with 16 B/lane coalesced float4 loads, wave(64)-per-row.
//  * ivf_scan reads each probed list's chunk exactly ONCE per query batch
//    (query tiles staged in LDS, looped inside the kernel), which makes the
//    algorithmic-bytes accounting the achievable ideal.
//  * Candidates and top-k entries are packed (monotone-mapped f32 key << 32
//    | row) into one u64 so selection is a single integer compare and ties
//    break deterministically toward the smaller row.
//  * Plain library GEMMs (query x centroid / query x database dots) go
//    through rocBLAS from the host side (dg_abi.cpp); everything irregular
//    is hand-written here.
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <cstdint>

#include "dg_internal.h"

#define WAVE 64

// ---------- float <-> order-preserving u32 ----------
__device__ __forceinline__ uint32_t enc_f32(float x) {
  uint32_t u = __float_as_uint(x);
  return (int32_t)u < 0 ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float dec_f32(uint32_t e) {
  uint32_t u = (e & 0x80000000u) ? (e & 0x7fffffffu) : ~e;
  return __uint_as_float(u);
}
__device__ __forceinline__ uint64_t pack_cand(float key, uint32_t row) {
  return ((uint64_t)enc_f32(key) << 32) | row;
}
static constexpr uint64_t kCandEmpty = ~0ull;

// ---------- small utilities ----------
__global__ void k_iota_i32(int32_t* p, int64_t n, int32_t v) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = v;
}

__global__ void k_row_norms(const float* __restrict__ x, int64_t n, int32_t d,
                            float* __restrict__ out) {
  // one wave per row, float4 coalesced
  int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  if (row >= n) return;
  const float* v = x + row * d;
  float acc = 0.f;
  int d4 = d / 4;
  const float4* v4 = (const float4*)v;
  for (int i = lane; i < d4; i += WAVE) {
    float4 a = v4[i];
    acc += a.x * a.x + a.y * a.y + a.z * a.z + a.w * a.w;
  }
  for (int i = d4 * 4 + lane; i < d; i += WAVE) acc += v[i] * v[i];
  for (int off = 32; off; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
  if (lane == 0) out[row] = acc;
}

__global__ void k_normalize_rows(float* __restrict__ x, int64_t n, int32_t d) {
  // NormalizeVectorForFaiss semantics (vector_index_utils.cc:480-491):
  // double-accumulated norm (distances_ref.cc:62-66), skip when
  // |1 - norm^2| <= 1e-5 or norm^2 == 0.
  int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  if (row >= n) return;
  float* v = x + row * d;
  double acc = 0.0;
  for (int i = lane; i < d; i += WAVE) acc += (double)v[i] * v[i];
  for (int off = 32; off; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
  float n2 = (float)acc;
  if (n2 > 0.f && fabsf(1.0f - n2) > 0.00001f) {
    float inv = 1.0f / sqrtf(n2);
    for (int i = lane; i < d; i += WAVE) v[i] = v[i] * inv;
  }
}

__global__ void k_argmin_rows(const float* __restrict__ dots,
                              const float* __restrict__ cnorms, int64_t n,
                              int32_t nlist, int metric,
                              int32_t* __restrict__ out) {
  // key: L2 = cnorm - 2*dot (qnorm constant per row); IP/COS = -dot.
  // tie: smaller list id (matches oracle strict '<').
  int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  if (row >= n) return;
  const float* dr = dots + row * nlist;
  uint64_t best = kCandEmpty;
  for (int l = lane; l < nlist; l += WAVE) {
    float key = (metric == 0) ? cnorms[l] - 2.0f * dr[l] : -dr[l];
    uint64_t c = pack_cand(key, (uint32_t)l);
    best = min(best, c);
  }
  for (int off = 32; off; off >>= 1)
    best = min(best, (uint64_t)__shfl_xor((long long)best, off, WAVE));
  if (lane == 0) out[row] = (int32_t)(best & 0xffffffffu);
}

// ---------- filter ----------
__device__ bool filter_pass(const dg_dev_filter f, int64_t id) {
  bool in;
  switch (f.kind) {
    case DG_FILTER_RANGE:
      in = (id >= f.min_id && id < f.max_id);  // vector_index.h:77-80
      break;
    case DG_FILTER_SORTED_IDS: {
      int64_t lo = 0, hi = f.n_ids - 1;
      in = false;
      while (lo <= hi) {  // SortFilterFunctor binary search,
        int64_t mid = (lo + hi) >> 1;  // vector_index.h:117-133
        int64_t v = f.ids[mid];
        if (v == id) { in = true; break; }
        if (id < v) hi = mid - 1; else lo = mid + 1;
      }
      break;
    }
    case DG_FILTER_BITMAP: {
      int64_t b = id - f.bitmap_base;
      in = (b >= 0 && b < f.bitmap_nbits) &&
           ((f.bitmap[b >> 6] >> (b & 63)) & 1);
      break;
    }
    default:
      in = true;
      return in;  // NONE: negate does not apply
  }
  return f.negate ? !in : in;
}

__global__ void k_build_pass_bitmap(const int64_t* __restrict__ ids, int64_t n,
                                    dg_dev_filter f,
                                    uint32_t* __restrict__ bitmap) {
  // bit per csr row: filter(id) && not tombstone (id >= 0)
  int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;  // word index
  int64_t nwords = (n + 31) / 32;
  if (w >= nwords) return;
  uint32_t bits = 0;
  for (int b = 0; b < 32; b++) {
    int64_t row = w * 32 + b;
    if (row < n) {
      int64_t id = ids[row];
      if (id >= 0 && filter_pass(f, id)) bits |= (1u << b);
    }
  }
  bitmap[w] = bits;
}

// ---------- generic block top-k over u64 candidates ----------
// Per-thread sorted ascending list in LDS, guarded insert, then ping-pong
// tree merge.  Dynamic LDS: 2 * T * k * 8 bytes.
__device__ void topk_insert(uint64_t* arr, int32_t& cnt, int32_t k,
                            uint64_t c) {
  if (cnt == k && c >= arr[k - 1]) return;
  int32_t pos = (cnt < k) ? cnt : k - 1;
  while (pos > 0 && arr[pos - 1] > c) {
    arr[pos] = arr[pos - 1];
    pos--;
  }
  arr[pos] = c;
  if (cnt < k) cnt++;
}

__device__ void topk_merge_block(uint64_t* lds, int32_t k, int32_t cnt,
                                 uint64_t* out_row /* k u64, thread 0 */) {
  // lds layout: [2][T][k]; each thread's list sorted asc, padded kCandEmpty
  const int T = blockDim.x;
  const int tid = threadIdx.x;
  uint64_t* a = lds;
  uint64_t* b = lds + (size_t)T * k;
  // pad own list
  for (int i = cnt; i < k; i++) a[(size_t)tid * k + i] = kCandEmpty;
  __syncthreads();
  int src_is_a = 1;
  for (int step = 1; step < T; step *= 2) {
    uint64_t* s = src_is_a ? a : b;
    uint64_t* d = src_is_a ? b : a;
    if ((tid & (2 * step - 1)) == 0 && tid + step < T) {
      const uint64_t* x = s + (size_t)tid * k;
      const uint64_t* y = s + (size_t)(tid + step) * k;
      uint64_t* o = d + (size_t)tid * k;
      int xi = 0, yi = 0;
      for (int i = 0; i < k; i++)
        o[i] = (yi >= k || (xi < k && x[xi] <= y[yi])) ? x[xi++] : y[yi++];
    } else if ((tid & (2 * step - 1)) == 0) {
      // no partner: copy through
      const uint64_t* x = s + (size_t)tid * k;
      uint64_t* o = d + (size_t)tid * k;
      for (int i = 0; i < k; i++) o[i] = x[i];
    }
    src_is_a ^= 1;
    __syncthreads();
  }
  if (tid == 0) {
    uint64_t* s = (src_is_a ? a : b);
    for (int i = 0; i < k; i++) out_row[i] = s[i];
  }
}

// dense scores row scan: mode 0 key=score, 1 key=cnorm[col]-2*score,
// 2 key=-score.  col_base added to the packed payload (and is the global
// column index for the bitmap); cnorms is indexed by the local column.
// ld = row stride of scores.  blockIdx.y picks a seg_w-wide column segment
// (the wide-k path runs (rows x S) blocks, writing k per segment at
// out_offset + y*k, merged afterwards with select_u64); plain callers
// launch gridDim.y = 1 with seg_w = cols.
__global__ void k_select_dense(const float* __restrict__ scores,
                               const float* __restrict__ cnorms, int64_t rows,
                               int64_t cols, int64_t ld, int64_t seg_w,
                               int32_t k, int mode,
                               const uint32_t* __restrict__ bitmap,
                               int64_t col_base,
                               uint64_t* __restrict__ out,
                               int64_t out_stride, int64_t out_offset) {
  extern __shared__ uint64_t lds[];
  int64_t row = blockIdx.x;
  if (row >= rows) return;
  const int64_t seg_start = (int64_t)blockIdx.y * seg_w;
  const int64_t cseg = min(seg_w, cols - seg_start);
  const float* sr = scores + row * ld + seg_start;
  const float* cn = cnorms + seg_start;
  uint64_t* mine = lds + (size_t)threadIdx.x * k;
  int32_t cnt = 0;
  for (int64_t c = threadIdx.x; c < cseg; c += blockDim.x) {
    if (bitmap) {
      int64_t g = col_base + seg_start + c;
      if (!((bitmap[g >> 5] >> (g & 31)) & 1)) continue;
    }
    float s = sr[c];
    float key = (mode == 0) ? s : (mode == 1) ? cn[c] - 2.0f * s : -s;
    topk_insert(mine, cnt, k,
                pack_cand(key, (uint32_t)(col_base + seg_start + c)));
  }
  topk_merge_block(lds, k, cnt,
                   out + row * out_stride + out_offset + blockIdx.y * k);
}

// candidate (u64) segment scan per query
__global__ void k_select_u64(const uint64_t* __restrict__ cand,
                             const int64_t* __restrict__ base,
                             const int64_t* __restrict__ total, int64_t nq,
                             int32_t k, uint64_t* __restrict__ out,
                             int64_t out_stride) {
  extern __shared__ uint64_t lds[];
  int64_t q = blockIdx.x;
  if (q >= nq) return;
  const uint64_t* seg = cand + (base ? base[q] : q * total[0]);
  int64_t len = base ? total[q] : total[0];
  uint64_t* mine = lds + (size_t)threadIdx.x * k;
  int32_t cnt = 0;
  for (int64_t i = threadIdx.x; i < len; i += blockDim.x)
    topk_insert(mine, cnt, k, seg[i]);
  topk_merge_block(lds, k, cnt, out + q * out_stride);
}

// emit: unpack sorted u64 topk -> faiss-convention distances + resolved ids
__global__ void k_emit(const uint64_t* __restrict__ topk,
                       const int64_t* __restrict__ ids_lookup,
                       const float* __restrict__ qnorms, int64_t nq, int32_t k,
                       int metric, int add_qnorm, float* __restrict__ out_dist,
                       int64_t* __restrict__ out_ids) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nq * k) return;
  int64_t q = i / k;
  uint64_t c = topk[i];
  if (c == kCandEmpty) {
    out_dist[i] = 0.0f;  // matches oracle topk_emit padding
    out_ids[i] = -1;
    return;
  }
  float key = dec_f32((uint32_t)(c >> 32));
  uint32_t row = (uint32_t)c;
  float dist;
  if (metric == 0) {  // L2: key may be cnorm-2dot; add qnorm back
    dist = add_qnorm ? key + qnorms[q] : key;
    if (dist < 0.f) dist = 0.f;  // clamp fp cancellation, faiss-style
  } else {
    dist = -key;  // raw IP score
  }
  out_dist[i] = dist;
  out_ids[i] = ids_lookup ? ids_lookup[row] : (int64_t)row;
}

// ---------- IVF probe machinery ----------
__global__ void k_probe_unpack(const uint64_t* __restrict__ topk, int64_t nq,
                               int32_t nprobe, const uint8_t* mask,
                               int32_t* __restrict__ probes) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nq * nprobe) return;
  uint64_t c = topk[i];
  int32_t l = (c == kCandEmpty) ? -1 : (int32_t)(c & 0xffffffffu);
  if (l >= 0 && mask && !mask[l]) l = -1;  // list-sharding ownership
  probes[i] = l;
}

__global__ void k_probes_all(int64_t nq, int32_t nprobe, const uint8_t* mask,
                             int32_t* __restrict__ probes) {
  // nprobe == nlist: probe every list in id order (no coarse selection)
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nq * nprobe) return;
  int32_t l = (int32_t)(i % nprobe);
  if (mask && !mask[l]) l = -1;
  probes[i] = l;
}

__global__ void k_hist_probes(const int32_t* __restrict__ probes, int64_t n,
                              int32_t* __restrict__ counts) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int32_t l = probes[i];
  if (l >= 0) atomicAdd(&counts[l], 1);
}

__global__ void k_scatter_probes(const int32_t* __restrict__ probes,
                                 int64_t nq, int32_t nprobe,
                                 int32_t* __restrict__ cursors,
                                 int32_t* __restrict__ inv_q,
                                 int32_t* __restrict__ inv_rank) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nq * nprobe) return;
  int32_t l = probes[i];
  if (l < 0) return;
  int32_t pos = atomicAdd(&cursors[l], 1);
  inv_q[pos] = (int32_t)(i / nprobe);
  inv_rank[pos] = (int32_t)(i % nprobe);
}

__global__ void k_cand_offsets(const int32_t* __restrict__ probes, int64_t nq,
                               int32_t nprobe,
                               const int64_t* __restrict__ csr_offsets,
                               int64_t* __restrict__ qp_off,
                               int64_t* __restrict__ q_total) {
  int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= nq) return;
  int64_t run = 0;
  for (int32_t p = 0; p < nprobe; p++) {
    qp_off[q * nprobe + p] = run;
    int32_t l = probes[q * nprobe + p];
    if (l >= 0) run += csr_offsets[l + 1] - csr_offsets[l];
  }
  q_total[q] = run;
}

__global__ void k_alg_bytes(const int32_t* __restrict__ inv_counts,
                            const int64_t* __restrict__ csr_offsets,
                            int32_t nlist, int64_t row_bytes,
                            int64_t* __restrict__ out) {
  __shared__ long long lds[256];
  long long acc = 0;
  for (int32_t l = blockIdx.x * blockDim.x + threadIdx.x; l < nlist;
       l += gridDim.x * blockDim.x)
    if (inv_counts[l] > 0)
      acc += (csr_offsets[l + 1] - csr_offsets[l]) * row_bytes;
  lds[threadIdx.x] = acc;
  __syncthreads();
  for (int s = blockDim.x / 2; s; s >>= 1) {
    if (threadIdx.x < s) lds[threadIdx.x] += lds[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0)
    atomicAdd((unsigned long long*)out, (unsigned long long)lds[0]);
}

__global__ void k_unit_counts(const int32_t* __restrict__ inv_counts,
                              int32_t nlist,
                              const int64_t* __restrict__ csr_offsets,
                              int32_t chunk_rows,
                              int32_t* __restrict__ unit_counts) {
  int32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l >= nlist) return;
  int64_t len = csr_offsets[l + 1] - csr_offsets[l];
  unit_counts[l] =
      (inv_counts[l] > 0 && len > 0) ? (int32_t)((len + chunk_rows - 1) / chunk_rows) : 0;
}

__global__ void k_fill_units(const int32_t* __restrict__ unit_offsets,
                             const int32_t* __restrict__ unit_counts,
                             int32_t nlist, uint32_t* __restrict__ units) {
  int32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l >= nlist) return;
  int32_t base = unit_offsets[l];
  for (int32_t c = 0; c < unit_counts[l]; c++) {
    units[2 * (base + c)] = (uint32_t)l;
    units[2 * (base + c) + 1] = (uint32_t)c;
  }
}

// ---------- range search (radius) ----------
// candidates (packed u64) below a per-query threshold key.  Strict '<'
// matches faiss RangeSearch (L2: dist < radius; IP: score > radius).
__global__ void k_count_below(const uint64_t* __restrict__ cand,
                              const int64_t* __restrict__ base,
                              const int64_t* __restrict__ total,
                              const uint64_t* __restrict__ thr, int64_t nq,
                              int64_t* __restrict__ counts) {
  __shared__ int64_t lds[256];
  int64_t q = blockIdx.x;
  if (q >= nq) return;
  const uint64_t* seg = cand + base[q];
  const uint64_t t = thr[q];
  int64_t c = 0;
  for (int64_t i = threadIdx.x; i < total[q]; i += blockDim.x)
    if (seg[i] < t) c++;
  lds[threadIdx.x] = c;
  __syncthreads();
  for (int s = blockDim.x / 2; s; s >>= 1) {
    if (threadIdx.x < s) lds[threadIdx.x] += lds[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) counts[q] = lds[0];
}

__global__ void k_compact_below(const uint64_t* __restrict__ cand,
                                const int64_t* __restrict__ base,
                                const int64_t* __restrict__ total,
                                const uint64_t* __restrict__ thr,
                                const int64_t* __restrict__ out_off,
                                int64_t nq, int64_t* __restrict__ cursors,
                                uint64_t* __restrict__ out) {
  int64_t q = blockIdx.x;
  if (q >= nq) return;
  const uint64_t* seg = cand + base[q];
  const uint64_t t = thr[q];
  for (int64_t i = threadIdx.x; i < total[q]; i += blockDim.x) {
    uint64_t c = seg[i];
    if (c < t) {
      int64_t pos = atomicAdd((unsigned long long*)&cursors[q], 1ull);
      out[out_off[q] + pos] = c;  // host sorts per query afterwards
    }
  }
}

// dense (Flat) variants over a dots chunk; mode as k_select_dense
__global__ void k_count_below_dense(const float* __restrict__ scores,
                                    const float* __restrict__ cnorms,
                                    int64_t rows, int64_t cols, int mode,
                                    const uint32_t* __restrict__ bitmap,
                                    int64_t col_base,
                                    const uint64_t* __restrict__ thr,
                                    int64_t* __restrict__ counts) {
  __shared__ int64_t lds[256];
  int64_t row = blockIdx.x;
  if (row >= rows) return;
  const float* sr = scores + row * cols;
  const uint64_t t = thr[row];
  int64_t c = 0;
  for (int64_t j = threadIdx.x; j < cols; j += blockDim.x) {
    int64_t g = col_base + j;
    if (bitmap && !((bitmap[g >> 5] >> (g & 31)) & 1)) continue;
    float s = sr[j];
    float key = (mode == 0) ? s : (mode == 1) ? cnorms[j] - 2.0f * s : -s;
    if (pack_cand(key, (uint32_t)g) < t) c++;
  }
  lds[threadIdx.x] = c;
  __syncthreads();
  for (int s2 = blockDim.x / 2; s2; s2 >>= 1) {
    if (threadIdx.x < s2) lds[threadIdx.x] += lds[threadIdx.x + s2];
    __syncthreads();
  }
  if (threadIdx.x == 0) counts[row] += lds[0];  // accumulates over chunks
}

__global__ void k_compact_below_dense(const float* __restrict__ scores,
                                      const float* __restrict__ cnorms,
                                      int64_t rows, int64_t cols, int mode,
                                      const uint32_t* __restrict__ bitmap,
                                      int64_t col_base,
                                      const uint64_t* __restrict__ thr,
                                      const int64_t* __restrict__ out_off,
                                      int64_t* __restrict__ cursors,
                                      uint64_t* __restrict__ out) {
  int64_t row = blockIdx.x;
  if (row >= rows) return;
  const float* sr = scores + row * cols;
  const uint64_t t = thr[row];
  for (int64_t j = threadIdx.x; j < cols; j += blockDim.x) {
    int64_t g = col_base + j;
    if (bitmap && !((bitmap[g >> 5] >> (g & 31)) & 1)) continue;
    float s = sr[j];
    float key = (mode == 0) ? s : (mode == 1) ? cnorms[j] - 2.0f * s : -s;
    uint64_t c = pack_cand(key, (uint32_t)g);
    if (c < t) {
      int64_t pos = atomicAdd((unsigned long long*)&cursors[row], 1ull);
      out[out_off[row] + pos] = c;
    }
  }
}

__global__ void k_range_emit(const uint64_t* __restrict__ packed,
                             const int64_t* __restrict__ lims,
                             const int64_t* __restrict__ ids_lookup,
                             const float* __restrict__ qnorms, int64_t nq,
                             int metric, int add_qnorm,
                             float* __restrict__ out_dist,
                             int64_t* __restrict__ out_ids) {
  int64_t q = blockIdx.x;
  if (q >= nq) return;
  for (int64_t i = lims[q] + threadIdx.x; i < lims[q + 1];
       i += blockDim.x) {
    uint64_t c = packed[i];
    float key = dec_f32((uint32_t)(c >> 32));
    uint32_t row = (uint32_t)c;
    float dist;
    if (metric == 0) {
      dist = add_qnorm ? key + qnorms[q] : key;
      if (dist < 0.f) dist = 0.f;
    } else {
      dist = -key;
    }
    out_dist[i] = dist;
    out_ids[i] = ids_lookup ? ids_lookup[row] : (int64_t)row;
  }
}

// ---------- misc small kernels ----------
__global__ void k_gather_rows_by_index(const float* __restrict__ src,
                                       const int64_t* __restrict__ idx,
                                       int64_t n, int32_t d,
                                       float* __restrict__ dst) {
  // dst[i] = src[idx[i]]  (wave per row)
  int64_t i = (int64_t)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  if (i >= n) return;
  const float4* s = (const float4*)(src + (size_t)idx[i] * d);
  float4* t = (float4*)(dst + (size_t)i * d);
  for (int j = lane; j < d / 4; j += WAVE) t[j] = s[j];
}

__global__ void k_fill_base_total(int64_t nq, int64_t len, int64_t* base,
                                  int64_t* total) {
  int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= nq) return;
  base[q] = q * len;
  total[q] = len;
}

__global__ void k_tombstone(const int64_t* __restrict__ pos, int64_t n,
                            int64_t* __restrict__ ids) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) ids[pos[i]] = -2;
}

// ---------- columnar scan (v2) ----------
// The dominant kernel, redesigned: CSR chunks are stored TRANSPOSED
// ([d][nrows_pad] per 1024-row chunk).  Each lane owns
// 4 consecutive rows and accumulates full dots for QTM LDS-staged queries
// in registers — no cross-lane reduction at all (the v1 wave-per-row form
// was bound by its 6-deep dependent shuffle chain per (row, query)).
// Column loads are 16 B/lane coalesced; query element broadcasts come from
// LDS.  Unused query slots are zero-staged so the inner loop is branch-free.
template <int QTM, int RPL, bool NT = false>  // queries/tile, rows/lane
__device__ __forceinline__ void ivf_scan_col_body(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  extern __shared__ __attribute__((aligned(16))) float smem[];  // [QTM * d]
  int64_t* cbase = (int64_t*)(smem + (size_t)QTM * d);  // [QTM]

  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* col = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];
  const int64_t row0 = list_start + (int64_t)chunk * chunk_rows;
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;

  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  __builtin_assume(d % 4 == 0 && d > 0);  // host enforces; enables b128 LDS

  for (int32_t t0 = 0; t0 < nql; t0 += QTM) {
    const int32_t qt = min(QTM, nql - t0);
    __syncthreads();
    for (int32_t j = 0; j < qt; j++) {
      int32_t q = inv_q[iq0 + t0 + j];
      const float4* src = (const float4*)(queries + (size_t)q * d);
      float4* dst = (float4*)(smem + (size_t)j * d);
      for (int i = threadIdx.x; i < d / 4; i += blockDim.x) dst[i] = src[i];
    }
    // zero-stage unused slots (keeps the inner loop branch-free)
    for (size_t i = (size_t)qt * d + threadIdx.x; i < (size_t)QTM * d;
         i += blockDim.x)
      smem[i] = 0.f;
    if (threadIdx.x < QTM) {
      int32_t j = threadIdx.x;
      if (j < qt) {
        int32_t q = inv_q[iq0 + t0 + j];
        int32_t rank = inv_rank[iq0 + t0 + j];
        cbase[j] = q_cand_base[q] + qp_off[(int64_t)q * nprobe + rank] -
                   list_start;
      } else {
        cbase[j] = 0;
      }
    }
    __syncthreads();

    // each wave covers WAVE*RPL consecutive rows; waves stride the chunk
    for (int32_t rb = wave_id * WAVE * RPL; rb < nrows_pad;
         rb += 4 * WAVE * RPL) {
      const int32_t rr0 = rb + lane * RPL;  // this lane's RPL rows
      if (rr0 >= nrows_pad) continue;
      float acc[QTM][RPL];
#pragma unroll
      for (int j = 0; j < QTM; j++)
#pragma unroll
        for (int x = 0; x < RPL; x++) acc[j][x] = 0.f;

      // dim loop unrolled by U, software-pipelined: while block i computes,
      // block i+U's loads are in flight (double-buffered column registers —
      // waiting within the issuing iteration exposes full HBM latency).
      // Query elements read 4-at-a-time from aligned LDS.
      constexpr int U = 4;
      static_assert(RPL == 1 || RPL == 2 || RPL % 4 == 0, "RPL");
      float ca[U][RPL], cb[U][RPL];
      const float* colp = col + rr0;
      auto load_block = [&](float (&c)[U][RPL], int32_t ib) {
#pragma unroll
        for (int u = 0; u < U; u++) {
          const float* cp = colp + (size_t)(ib + u) * nrows_pad;
          if (RPL >= 4) {
#pragma unroll
            for (int v = 0; v < RPL / 4; v++) {
              // nt (stream-once) column loads selectable: each chunk is
              // read by exactly one CU per batch, so L1/L2 retention buys
              // nothing (MI355X_MICROARCH.md nt-weights row)
              typedef float nf4 __attribute__((ext_vector_type(4)));
              float4 c4;
              if (NT) {
                const nf4 nv =
                    __builtin_nontemporal_load((const nf4*)cp + v);
                c4 = make_float4(nv[0], nv[1], nv[2], nv[3]);
              } else {
                c4 = ((const float4*)cp)[v];
              }
              c[u][4 * v + 0] = c4.x;
              c[u][4 * v + 1] = c4.y;
              c[u][4 * v + 2] = c4.z;
              c[u][4 * v + 3] = c4.w;
            }
          } else if (RPL == 2) {
            const float2 c2 = *(const float2*)cp;
            c[u][0] = c2.x; c[u][RPL - 1] = c2.y;
          } else {
            c[u][0] = *cp;
          }
        }
      };
      static_assert(U % 4 == 0, "U");
      auto compute_block = [&](float (&c)[U][RPL], int32_t ib) {
#pragma unroll
        for (int j = 0; j < QTM; j++) {
#pragma unroll
          for (int u4 = 0; u4 < U / 4; u4++) {
            // two b64 LDS reads (2cy/8B) instead of a float4 that lowers
            // to ds_read2_b32 (4cy/8B)
            const float* qp = smem + (size_t)j * d + ib + u4 * 4;
            const float2 qa = *(const float2*)qp;
            const float2 qb = *(const float2*)(qp + 2);
            const float qv[4] = {qa.x, qa.y, qb.x, qb.y};
#pragma unroll
            for (int uu = 0; uu < 4; uu++)
#pragma unroll
              for (int x = 0; x < RPL; x++)
                acc[j][x] += c[u4 * 4 + uu][x] * qv[uu];
          }
        }
      };
      // 2-deep rotation (3-deep costs a wave of occupancy and regresses)
      load_block(ca, 0);
      int32_t ib = 0;
      for (; ib + U < d; ib += 2 * U) {
        load_block(cb, ib + U);
        compute_block(ca, ib);
        if (ib + 2 * U < d) load_block(ca, ib + 2 * U);
        compute_block(cb, ib + U);
      }
      if (ib < d) compute_block(ca, ib);

      // emit: lane's rows are consecutive in each query's segment
      // (guards, not breaks: a break blocks full unroll and acc[] would be
      // dynamically indexed -> scratch spill)
#pragma unroll
      for (int j = 0; j < QTM; j++) {
        if (j < qt) {
          const int64_t cb = cbase[j] + row0 + rr0;
#pragma unroll
          for (int x = 0; x < RPL; x++) {
            const int32_t rl = rr0 + x;
            if (rl < nrows) {
              const int64_t r = row0 + rl;
              bool pass = true;
              if (bitmap) pass = (bitmap[r >> 5] >> (r & 31)) & 1;
              float key = (metric == 0) ? vnorms[r] - 2.0f * acc[j][x]
                                        : -acc[j][x];
              cand[cb + x] = pass ? pack_cand(key, (uint32_t)r) : kCandEmpty;
            }
          }
        }
      }
    }
  }
}

// ---------- IVF-PQ kernels ----------
// ADC decomposition: for L2,
//   ||q - (c_l + r)||^2 = ||q||^2 - 2*dots[q][l]
//                         + sum_m ( S[l][m][code] - 2*T[q][m][code] )
// with S[l][m][c] = ||centroid_l_sub_m + codebook_m[c]||^2 (index-static)
// and  T[q][m][c] = q_sub_m . codebook_m[c] (per batch, strided GEMM).
// For IP: score = dots[q][l] + sum_m T[q][m][code].
// Restates faiss IndexIVFPQ residual ADC (oracle.c dgo_ivfpq_search is the
// direct restatement; this is the same arithmetic regrouped).

__global__ void k_residual(const float* __restrict__ x,
                           const int32_t* __restrict__ assign,
                           const float* __restrict__ centroids, int64_t n,
                           int32_t d, float* __restrict__ out) {
  int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  if (row >= n) return;
  const float* v = x + (size_t)row * d;
  const float* c = centroids + (size_t)assign[row] * d;
  float* o = out + (size_t)row * d;
  for (int i = lane; i < d; i += WAVE) o[i] = v[i] - c[i];
}

__global__ void k_set_code(const int32_t* __restrict__ amin, int64_t n,
                           int32_t m, int32_t M, uint8_t* __restrict__ codes) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) codes[(size_t)i * M + m] = (uint8_t)amin[i];
}

__global__ void k_gather_codes(const uint8_t* __restrict__ src,
                               const uint32_t* __restrict__ perm, int64_t n,
                               int32_t M, uint8_t* __restrict__ dst) {
  // dst[perm[i]] = src[i], M bytes per row, dword copies
  int64_t i = (int64_t)blockIdx.x * (blockDim.x / 32) + threadIdx.x / 32;
  int sub = threadIdx.x % 32;
  if (i >= n) return;
  const uint32_t* s = (const uint32_t*)(src + (size_t)i * M);
  uint32_t* t = (uint32_t*)(dst + (size_t)perm[i] * M);
  for (int j = sub; j < M / 4; j += 32) t[j] = s[j];
}

__global__ void k_build_S(const float* __restrict__ centroids,
                          const float* __restrict__ codebooks, int32_t nlist,
                          int32_t M, int32_t dsub, int32_t d,
                          __half* __restrict__ S) {
  // one thread per (l, m, code): ||c_sub + cb||^2 over dsub elems
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)nlist * M * 256;
  if (i >= total) return;
  int32_t code = (int32_t)(i & 255);
  int32_t m = (int32_t)((i >> 8) % M);
  int32_t l = (int32_t)(i / (256 * M));
  const float* c = centroids + (size_t)l * d + m * dsub;
  const float* cb = codebooks + ((size_t)m * 256 + code) * dsub;
  float acc = 0.f;
  for (int32_t j = 0; j < dsub; j++) {
    float t = c[j] + cb[j];
    acc += t * t;
  }
  S[i] = __float2half(acc);
}

__global__ void k_f32_to_f16(const float* __restrict__ in, int64_t n,
                             __half* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = __float2half(in[i]);
}

// THE PQ scan: unit = (list, chunk of codes).  Codes for a RPV*64-vector tile
// are LDS-staged once and reused by every probing query; each LANE owns one
// vector and walks m = 0..M-1 sequentially, so all lanes of a wave gather
// within the same 1 KB rows of T[q] and S[l] (L1-resident after first
// touch).  No cross-lane reduction.
// COOP: all 4 waves walk the SAME query over disjoint quarters of the
// staged tile (one coherent 48 KB T-window in L1 at a time) instead of 4
// different queries (4 windows thrash the 32 KB L1).
template <int RPV, int MINB = 1, bool COOP = false>
__global__ void __launch_bounds__(256, MINB) k_ivfpq_scan(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const uint8_t* __restrict__ csr_codes, const __half* __restrict__ S,
    const __half* __restrict__ T, const float* __restrict__ coarse_dots,
    int32_t nlist, int32_t M, const int32_t* __restrict__ inv_offsets,
    const int32_t* __restrict__ inv_q, const int32_t* __restrict__ inv_rank,
    const int64_t* __restrict__ qp_off, const int64_t* __restrict__ q_cand_base,
    int32_t nprobe, int metric, const uint32_t* __restrict__ bitmap,
    int32_t chunk_rows, uint64_t* __restrict__ cand) {
  extern __shared__ __attribute__((aligned(16))) uint8_t lds_codes[];
  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t list_end = csr_offsets[list + 1];
  const int64_t row_start = list_start + (int64_t)chunk * chunk_rows;
  const int64_t row_end = min(list_end, row_start + chunk_rows);
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;
  if (nql == 0) return;  // unprobed list (all-chunks launch)
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int32_t TILE = RPV * WAVE;  // vectors staged per pass

  // stage S_l into LDS once per unit (M*256 f16 = 48 KB at M=96); without
  // this every (wave, query, m) visit re-fetches a 512 B row from L2/HBM -
  // S exceeds L3 at cfg D and those fetches were the kernel's bound
  __half* lds_S = (__half*)(lds_codes + (size_t)TILE * M);
  {
    const uint32_t* src_s = (const uint32_t*)(S + (size_t)list * M * 256);
    uint32_t* dst_s = (uint32_t*)lds_S;
    const int32_t words = M * 256 / 2;
    for (int32_t w = threadIdx.x; w < words; w += blockDim.x)
      dst_s[w] = src_s[w];
  }
  __syncthreads();
  const __half* Sl = lds_S;

  for (int64_t t0 = row_start; t0 < row_end; t0 += TILE) {
    const int32_t tn = (int32_t)min((int64_t)TILE, row_end - t0);
    __syncthreads();
    {  // stage codes coalesced (rows are contiguous in csr_codes)
      const uint32_t* src = (const uint32_t*)(csr_codes + (size_t)t0 * M);
      uint32_t* dst = (uint32_t*)lds_codes;
      int32_t words = tn * M / 4;
      for (int32_t w = threadIdx.x; w < words; w += blockDim.x)
        dst[w] = src[w];
    }
    __syncthreads();

    // waves split the probing queries; each LANE owns RPV of the staged
    // vectors, so every (wave, query, m) visit of a 1 KB table row is
    // consumed by the whole tile — without this, 3 KB of S/T row traffic
    // per (vector, query) pair is the bottleneck (S at cfg D is 1.6 GB,
    // beyond L3).  COOP instead walks queries jointly (see above).
    constexpr int RV = COOP ? RPV / 4 : RPV;
    const int32_t vbase = COOP ? wave_id * (TILE / 4) : 0;
    for (int32_t qi = COOP ? 0 : wave_id; qi < nql;
         qi += COOP ? 1 : blockDim.x / WAVE) {
      const int32_t q = inv_q[iq0 + qi];
      const int32_t rank = inv_rank[iq0 + qi];
      const __half* Tq = T + (size_t)q * M * 256;
      const float dot = coarse_dots[(size_t)q * nlist + list];
      const int64_t cb0 = q_cand_base[q] +
                          qp_off[(int64_t)q * nprobe + rank] - list_start;
      float acc[RV];
      const uint32_t* code4[RV];
#pragma unroll
      for (int r = 0; r < RV; r++) {
        acc[r] = 0.f;
        code4[r] = (const uint32_t*)(lds_codes +
                                     (size_t)(vbase + lane + r * WAVE) * M);
      }
      if (metric == 0) {
        const __half* Sm = Sl;
        const __half* Tm = Tq;
        for (int32_t m4 = 0; m4 < M / 4; m4++) {
#pragma unroll
          for (int r = 0; r < RV; r++) {
            const uint32_t cw = code4[r][m4];
            const uint32_t c0 = cw & 255, c1 = (cw >> 8) & 255,
                           c2 = (cw >> 16) & 255, c3 = cw >> 24;
            acc[r] += (__half2float(Sm[c0]) + __half2float(Sm[256 + c1]) +
                       __half2float(Sm[512 + c2]) +
                       __half2float(Sm[768 + c3])) -
                      2.0f * (__half2float(Tm[c0]) +
                              __half2float(Tm[256 + c1]) +
                              __half2float(Tm[512 + c2]) +
                              __half2float(Tm[768 + c3]));
          }
          Sm += 1024;
          Tm += 1024;
        }
#pragma unroll
        for (int r = 0; r < RV; r++) acc[r] -= 2.0f * dot;  // +qnorm@emit
      } else {
        const __half* Tm = Tq;
        for (int32_t m4 = 0; m4 < M / 4; m4++) {
#pragma unroll
          for (int r = 0; r < RV; r++) {
            const uint32_t cw = code4[r][m4];
            acc[r] += __half2float(Tm[cw & 255]) +
                      __half2float(Tm[256 + ((cw >> 8) & 255)]) +
                      __half2float(Tm[512 + ((cw >> 16) & 255)]) +
                      __half2float(Tm[768 + (cw >> 24)]);
          }
          Tm += 1024;
        }
#pragma unroll
        for (int r = 0; r < RV; r++) acc[r] = -(acc[r] + dot);
      }
#pragma unroll
      for (int r = 0; r < RV; r++) {
        const int32_t v = vbase + lane + r * WAVE;
        if (v < tn) {
          const int64_t rw = t0 + v;
          bool pass = true;
          if (bitmap) pass = (bitmap[rw >> 5] >> (rw & 31)) & 1;
          cand[cb0 + rw] =
              pass ? pack_cand(acc[r], (uint32_t)rw) : kCandEmpty;
        }
      }
    }
  }
}

// ---------- asm-pipelined columnar scan (v3, DG_SCAN_VARIANT=10) ----------
// Round-2 redesign of the dim loop.  The C++ double-buffer rotation of v2
// compiles to a register-allocator artifact: hipcc copies each loaded
// float4 into the loop-carried phi registers THROUGH an s_waitcnt vmcnt(0)
// placed right after the first load of the next block — the "pipeline"
// waits for the load it just issued, exposing full HBM latency (~900 cy)
// every 8 dims.  Here the loads are inline-asm global_load_dwordx4 into
// explicitly tied buffers with counted s_waitcnt vmcnt(N) (the compiler
// does not know the loads are pending, so it cannot insert its own waits),
// modulo-scheduled over DEPTH=4 stages x U=4 dims: 16 column loads stay in
// flight per lane while one stage computes — outstanding bytes per CU
// (8 waves x ~12 KB) far exceed the ~7 KB Little's-law need at the 6.3 TB/s
// achievable ceiling.  vmcnt discipline: the dim loop contains NO other
// vmem (a vmcnt(0) drain precedes the prologue so emit-stores of the
// previous tile don't skew the counts); loads past d land in d_csr_t's
// one-chunk slack and are never consumed.
typedef float dg_f4 __attribute__((ext_vector_type(4)));

// NOTE: the address must be bound as uint64_t — a pointer-typed "v"
// operand binds a single 32-bit VGPR and truncates the VA (measured: GPU
// memory fault at the truncated low-32-bit address).  "=&v" (early
// clobber) keeps the allocator from overlapping the destination tuple
// with the address pair: the hardware does not interlock VALU writes to
// an ASYNC load's destination, and the compiler cannot see that the asm
// load is still in flight.
#define DG_GLOAD4(dst, ptr)                                         \
  asm volatile("global_load_dwordx4 %0, %1, off"                    \
               : "=&v"(dst)                                         \
               : "v"((uint64_t)(uintptr_t)(ptr)))
// counted wait, tying the stage's 4 buffers so their consumers cannot be
// scheduled above the wait.  SAFE mode drains everything (diagnostic
// baseline for the counted schedule).
#define DG_WAITV(n, b)                                              \
  asm volatile("s_waitcnt vmcnt(" #n ")"                            \
               : "+v"((b)[0]), "+v"((b)[1]), "+v"((b)[2]), "+v"((b)[3]))
#define DG_WAITV_N(n, b)                                            \
  do {                                                              \
    if (SAFE)                                                       \
      DG_WAITV(0, b);                                               \
    else                                                            \
      DG_WAITV(n, b);                                               \
  } while (0)

// row-scan body of k_ivf_scan_pipe, templated on the EFFECTIVE query-tile
// width JT: tiles that fill at most half the QTM slots take the JT=QTM/2
// instantiation and skip the zero-padded FMAs (at cfg C the mean tile
// holds ~8 of 16 queries — half the compute multiplies zeros otherwise).
template <int JT, bool SAFE>
__device__ __forceinline__ void dg_scan_rows_body(
    const float* smem, const int64_t* cbase, const float* col,
    const float* __restrict__ vnorms, int32_t d, int32_t nrows,
    int32_t nrows_pad, int32_t rr0, int64_t row0, int32_t qt, int metric,
    const uint32_t* __restrict__ bitmap, uint64_t* __restrict__ cand) {
  constexpr int RPL = 4;
  constexpr int U = 4;
  float acc[JT][RPL];
#pragma unroll
  for (int j = 0; j < JT; j++)
#pragma unroll
    for (int x = 0; x < RPL; x++) acc[j][x] = 0.f;

  // drain prior vmem (previous tile's emit stores) so vmcnt counts below
  // track ONLY this loop's column loads
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const char* colp = (const char*)(col + rr0);
  const size_t cstride = (size_t)nrows_pad * 4;  // bytes per dim column
  dg_f4 b0[U], b1[U], b2[U], b3[U];
  auto issue = [&](dg_f4 (&b)[U], int32_t ib) {
    const char* p = colp + (size_t)ib * cstride;
#pragma unroll
    for (int u = 0; u < U; u++) DG_GLOAD4(b[u], p + u * cstride);
  };
  auto compute = [&](dg_f4 (&b)[U], int32_t ib) {
#pragma unroll
    for (int j = 0; j < JT; j++) {
      const float* qp = smem + (size_t)j * d + ib;
      const float2 qa = *(const float2*)qp;
      const float2 qb = *(const float2*)(qp + 2);
      const float qv[4] = {qa.x, qa.y, qb.x, qb.y};
#pragma unroll
      for (int u = 0; u < U; u++)
#pragma unroll
        for (int x = 0; x < RPL; x++) acc[j][x] += b[u][x] * qv[u];
    }
  };
  // modulo schedule, period 4 stages = 16 dims
  issue(b0, 0);
  issue(b1, U);
  issue(b2, 2 * U);
  issue(b3, 3 * U);
  int32_t base = 0;
  for (; base + 16 < d; base += 16) {
    DG_WAITV_N(12, b0);
    compute(b0, base);
    issue(b0, base + 16);
    DG_WAITV_N(12, b1);
    compute(b1, base + 4);
    issue(b1, base + 20);
    DG_WAITV_N(12, b2);
    compute(b2, base + 8);
    issue(b2, base + 24);
    DG_WAITV_N(12, b3);
    compute(b3, base + 12);
    issue(b3, base + 28);
  }
  {  // epilogue: r = d - base in {4, 8, 12, 16} (d % 4 == 0)
    const int32_t r = d - base;
    DG_WAITV_N(12, b0);
    compute(b0, base);
    if (r > 4) {
      DG_WAITV_N(8, b1);
      compute(b1, base + 4);
    }
    if (r > 8) {
      DG_WAITV_N(4, b2);
      compute(b2, base + 8);
    }
    if (r > 12) {
      DG_WAITV(0, b3);
      compute(b3, base + 12);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

#pragma unroll
  for (int j = 0; j < JT; j++) {
    if (j < qt) {
      const int64_t cb = cbase[j] + row0 + rr0;
#pragma unroll
      for (int x = 0; x < RPL; x++) {
        const int32_t rl = rr0 + x;
        if (rl < nrows) {
          const int64_t r = row0 + rl;
          bool pass = true;
          if (bitmap) pass = (bitmap[r >> 5] >> (r & 31)) & 1;
          float key = (metric == 0) ? vnorms[r] - 2.0f * acc[j][x]
                                    : -acc[j][x];
          cand[cb + x] = pass ? pack_cand(key, (uint32_t)r) : kCandEmpty;
        }
      }
    }
  }
}

// v6 (variant 19): two banks x 8 dims, ordered wait(A) -> issue(B_next)
// -> compute(A): the other bank's loads are issued a FULL 8-dim compute
// (~1000 cy) before their drain, so each drain should find them landed.
// Same full-drain tie discipline as the shipped pipeline.
template <int JT>
__device__ __forceinline__ void dg_scan_rows_body2(
    const float* smem, const int64_t* cbase, const float* col,
    const float* __restrict__ vnorms, int32_t d, int32_t nrows,
    int32_t nrows_pad, int32_t rr0, int64_t row0, int32_t qt, int metric,
    const uint32_t* __restrict__ bitmap, uint64_t* __restrict__ cand) {
  constexpr int RPL = 4;
  float acc[JT][RPL];
#pragma unroll
  for (int j = 0; j < JT; j++)
#pragma unroll
    for (int x = 0; x < RPL; x++) acc[j][x] = 0.f;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  const char* colp = (const char*)(col + rr0);
  const size_t cstride = (size_t)nrows_pad * 4;
  dg_f4 A[8], B[8];
  auto issue8 = [&](dg_f4 (&b)[8], int32_t ib) {
#pragma unroll
    for (int u = 0; u < 8; u++)
      DG_GLOAD4(b[u], colp + (size_t)(ib + u) * cstride);
  };
  auto wait8 = [&](dg_f4 (&b)[8]) {
    asm volatile("s_waitcnt vmcnt(0)"
                 : "+v"(b[0]), "+v"(b[1]), "+v"(b[2]), "+v"(b[3]),
                   "+v"(b[4]), "+v"(b[5]), "+v"(b[6]), "+v"(b[7]));
  };
  auto compute8 = [&](dg_f4 (&b)[8], int32_t ib) {
#pragma unroll
    for (int j = 0; j < JT; j++) {
      const float* qp = smem + (size_t)j * d + ib;
#pragma unroll
      for (int h = 0; h < 4; h++) {
        const float2 qq = *(const float2*)(qp + 2 * h);
#pragma unroll
        for (int x = 0; x < RPL; x++)
          acc[j][x] += b[2 * h][x] * qq.x + b[2 * h + 1][x] * qq.y;
      }
    }
  };
  int32_t base = 0;
  if (d >= 16) {
    issue8(A, 0);
    for (; base + 16 <= d; base += 16) {
      wait8(A);
      if (base + 24 <= d) issue8(B, base + 8);
      compute8(A, base);
      if (base + 24 <= d) {
        wait8(B);
        if (base + 32 <= d) issue8(A, base + 16);
        compute8(B, base + 8);
      } else {
        // last 8 dims of the main region handled by the tail below
        base += 8;
        break;
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  const float* colf = col + rr0;
  for (int32_t ib = base; ib < d; ib += 4) {
    float4 c[4];
#pragma unroll
    for (int u = 0; u < 4; u++)
      c[u] = (ib + u < d)
                 ? *(const float4*)(colf + (size_t)(ib + u) * nrows_pad)
                 : float4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int j = 0; j < JT; j++) {
      const float* qp = smem + (size_t)j * d + ib;
#pragma unroll
      for (int u = 0; u < 4 && ib + u < d; u++) {
        const float qv = qp[u];
        acc[j][0] += c[u].x * qv;
        acc[j][1] += c[u].y * qv;
        acc[j][2] += c[u].z * qv;
        acc[j][3] += c[u].w * qv;
      }
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
  for (int j = 0; j < JT; j++) {
    if (j < qt) {
      const int64_t cb = cbase[j] + row0 + rr0;
#pragma unroll
      for (int x = 0; x < RPL; x++) {
        const int32_t rl = rr0 + x;
        if (rl < nrows) {
          const int64_t r = row0 + rl;
          bool pass = true;
          if (bitmap) pass = (bitmap[r >> 5] >> (r & 31)) & 1;
          float key = (metric == 0) ? vnorms[r] - 2.0f * acc[j][x]
                                    : -acc[j][x];
          cand[cb + x] = pass ? pack_cand(key, (uint32_t)r) : kCandEmpty;
        }
      }
    }
  }
}

template <int QTM, bool SAFE = false>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_pipe(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  constexpr int RPL = 4;  // rows per lane (float4 column loads)
  extern __shared__ __attribute__((aligned(16))) float smem[];  // [QTM * d]
  int64_t* cbase = (int64_t*)(smem + (size_t)QTM * d);          // [QTM]

  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* col = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];
  const int64_t row0 = list_start + (int64_t)chunk * chunk_rows;
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;

  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  __builtin_assume(d % 4 == 0 && d > 0);

  for (int32_t t0 = 0; t0 < nql; t0 += QTM) {
    const int32_t qt = min(QTM, nql - t0);
    __syncthreads();
    for (int32_t j = 0; j < qt; j++) {
      int32_t q = inv_q[iq0 + t0 + j];
      const float4* src = (const float4*)(queries + (size_t)q * d);
      float4* dst = (float4*)(smem + (size_t)j * d);
      for (int i = threadIdx.x; i < d / 4; i += blockDim.x) dst[i] = src[i];
    }
    for (size_t i = (size_t)qt * d + threadIdx.x; i < (size_t)QTM * d;
         i += blockDim.x)
      smem[i] = 0.f;
    if (threadIdx.x < QTM) {
      int32_t j = threadIdx.x;
      if (j < qt) {
        int32_t q = inv_q[iq0 + t0 + j];
        int32_t rank = inv_rank[iq0 + t0 + j];
        cbase[j] = q_cand_base[q] + qp_off[(int64_t)q * nprobe + rank] -
                   list_start;
      } else {
        cbase[j] = 0;
      }
    }
    __syncthreads();

    for (int32_t rb = wave_id * WAVE * RPL; rb < nrows_pad;
         rb += 4 * WAVE * RPL) {
      const int32_t rr0 = rb + lane * RPL;
      if (rr0 >= nrows_pad) continue;
      if (QTM >= 16 && qt <= QTM / 2)
        dg_scan_rows_body<(QTM >= 16 ? QTM / 2 : QTM), SAFE>(
            smem, cbase, col, vnorms, d, nrows, nrows_pad, rr0, row0, qt,
            metric, bitmap, cand);
      else
        dg_scan_rows_body<QTM, SAFE>(smem, cbase, col, vnorms, d, nrows,
                                     nrows_pad, rr0, row0, qt, metric,
                                     bitmap, cand);
    }
  }
}

// ---------- generalized drain pipeline (v5, DG_SCAN_VARIANT=16..18) ----
// SAFE-mode generalization of k_ivf_scan_pipe: S stages x 4 dims issued as
// one batch at block end, ONE vmcnt(0) drain at the next block's head
// (tied per stage before its compute), amortizing the exposed HBM latency
// over 4*S dims instead of 4.  All waits are full drains, so the LLVM
// in-flight-register hazards of the counted variants cannot occur.
template <int QTM, int S>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_pipe3(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  constexpr int RPL = 4;
  extern __shared__ __attribute__((aligned(16))) float smem[];  // [QTM * d]
  int64_t* cbase = (int64_t*)(smem + (size_t)QTM * d);          // [QTM]

  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* col = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];
  const int64_t row0 = list_start + (int64_t)chunk * chunk_rows;
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;

  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  __builtin_assume(d % 4 == 0 && d > 0);

  for (int32_t t0 = 0; t0 < nql; t0 += QTM) {
    const int32_t qt = min(QTM, nql - t0);
    __syncthreads();
    for (int32_t j = 0; j < qt; j++) {
      int32_t q = inv_q[iq0 + t0 + j];
      const float4* src = (const float4*)(queries + (size_t)q * d);
      float4* dst = (float4*)(smem + (size_t)j * d);
      for (int i = threadIdx.x; i < d / 4; i += blockDim.x) dst[i] = src[i];
    }
    for (size_t i = (size_t)qt * d + threadIdx.x; i < (size_t)QTM * d;
         i += blockDim.x)
      smem[i] = 0.f;
    if (threadIdx.x < QTM) {
      int32_t j = threadIdx.x;
      if (j < qt) {
        int32_t q = inv_q[iq0 + t0 + j];
        int32_t rank = inv_rank[iq0 + t0 + j];
        cbase[j] = q_cand_base[q] + qp_off[(int64_t)q * nprobe + rank] -
                   list_start;
      } else {
        cbase[j] = 0;
      }
    }
    __syncthreads();

    for (int32_t rb = wave_id * WAVE * RPL; rb < nrows_pad;
         rb += 4 * WAVE * RPL) {
      const int32_t rr0 = rb + lane * RPL;
      if (rr0 >= nrows_pad) continue;
      float acc[QTM][RPL];
#pragma unroll
      for (int j = 0; j < QTM; j++)
#pragma unroll
        for (int x = 0; x < RPL; x++) acc[j][x] = 0.f;

      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // prior emit stores
      const char* colp0 = (const char*)(col + rr0);
      const size_t cstride = (size_t)nrows_pad * 4;
      dg_f4 buf[S][4];
      auto issue_block = [&](int32_t ib) {
#pragma unroll
        for (int i = 0; i < S; i++)
#pragma unroll
          for (int u = 0; u < 4; u++)
            DG_GLOAD4(buf[i][u],
                      colp0 + (size_t)(ib + 4 * i + u) * cstride);
      };
      auto compute4 = [&](const dg_f4 (&b)[4], int32_t ib) {
#pragma unroll
        for (int j = 0; j < QTM; j++) {
          const float* qp = smem + (size_t)j * d + ib;
          const float2 qa = *(const float2*)qp;
          const float2 qb = *(const float2*)(qp + 2);
          const float qv[4] = {qa.x, qa.y, qb.x, qb.y};
#pragma unroll
          for (int u = 0; u < 4; u++)
#pragma unroll
            for (int x = 0; x < RPL; x++) acc[j][x] += b[u][x] * qv[u];
        }
      };
      constexpr int BD = 4 * S;  // dims per block
      int32_t base = 0;
      if (d >= BD) {
        issue_block(0);
        for (; base + BD <= d; base += BD) {
          const bool more = base + 2 * BD <= d;
#pragma unroll
          for (int i = 0; i < S; i++) {
            DG_WAITV(0, buf[i]);  // first drains everything; rest order
            compute4(buf[i], base + 4 * i);
          }
          if (more) issue_block(base + BD);
        }
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // tail < BD dims: compiler-managed loads
      const float* colf = col + rr0;
      for (int32_t ib = base; ib < d; ib += 4) {
        float4 c[4];
#pragma unroll
        for (int u = 0; u < 4; u++)
          c[u] = (ib + u < d)
                     ? *(const float4*)(colf + (size_t)(ib + u) * nrows_pad)
                     : float4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int j = 0; j < QTM; j++) {
          const float* qp = smem + (size_t)j * d + ib;
#pragma unroll
          for (int u = 0; u < 4 && ib + u < d; u++) {
            const float qv = qp[u];
            acc[j][0] += c[u].x * qv;
            acc[j][1] += c[u].y * qv;
            acc[j][2] += c[u].z * qv;
            acc[j][3] += c[u].w * qv;
          }
        }
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

#pragma unroll
      for (int j = 0; j < QTM; j++) {
        if (j < qt) {
          const int64_t cb = cbase[j] + row0 + rr0;
#pragma unroll
          for (int x = 0; x < RPL; x++) {
            const int32_t rl = rr0 + x;
            if (rl < nrows) {
              const int64_t r = row0 + rl;
              bool pass = true;
              if (bitmap) pass = (bitmap[r >> 5] >> (r & 31)) & 1;
              float key = (metric == 0) ? vnorms[r] - 2.0f * acc[j][x]
                                        : -acc[j][x];
              cand[cb + x] = pass ? pack_cand(key, (uint32_t)r) : kCandEmpty;
            }
          }
        }
      }
    }
  }
}


template <int QTM>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_pipe4(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  constexpr int RPL = 4;
  extern __shared__ __attribute__((aligned(16))) float smem[];
  int64_t* cbase = (int64_t*)(smem + (size_t)QTM * d);
  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* col = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];
  const int64_t row0 = list_start + (int64_t)chunk * chunk_rows;
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  __builtin_assume(d % 4 == 0 && d > 0);
  for (int32_t t0 = 0; t0 < nql; t0 += QTM) {
    const int32_t qt = min(QTM, nql - t0);
    __syncthreads();
    for (int32_t j = 0; j < qt; j++) {
      int32_t q = inv_q[iq0 + t0 + j];
      const float4* src = (const float4*)(queries + (size_t)q * d);
      float4* dst = (float4*)(smem + (size_t)j * d);
      for (int i = threadIdx.x; i < d / 4; i += blockDim.x) dst[i] = src[i];
    }
    for (size_t i = (size_t)qt * d + threadIdx.x; i < (size_t)QTM * d;
         i += blockDim.x)
      smem[i] = 0.f;
    if (threadIdx.x < QTM) {
      int32_t j = threadIdx.x;
      if (j < qt) {
        int32_t q = inv_q[iq0 + t0 + j];
        int32_t rank = inv_rank[iq0 + t0 + j];
        cbase[j] = q_cand_base[q] + qp_off[(int64_t)q * nprobe + rank] -
                   list_start;
      } else {
        cbase[j] = 0;
      }
    }
    __syncthreads();
    for (int32_t rb = wave_id * WAVE * RPL; rb < nrows_pad;
         rb += 4 * WAVE * RPL) {
      const int32_t rr0 = rb + lane * RPL;
      if (rr0 >= nrows_pad) continue;
      if (QTM >= 16 && qt <= QTM / 2)
        dg_scan_rows_body2<(QTM >= 16 ? QTM / 2 : QTM)>(
            smem, cbase, col, vnorms, d, nrows, nrows_pad, rr0, row0, qt,
            metric, bitmap, cand);
      else
        dg_scan_rows_body2<QTM>(smem, cbase, col, vnorms, d, nrows,
                                nrows_pad, rr0, row0, qt, metric, bitmap,
                                cand);
    }
  }
}

// ---------- fused-stage asm scan (v4, DG_SCAN_VARIANT=13) ----------
// The v3 kernel's separate issue/wait asms are not airtight: LLVM may copy
// or reuse an in-flight load's destination registers between the asm
// statements (it models asm loads as completing instantly; verified in the
// emitted code by a static hazard scan).  Here each 8-dim stage is ONE asm
// block — 8 global_load_dwordx4 into the next bank, address bumps, and the
// counted s_waitcnt vmcnt(8) that retires the CURRENT bank — so no
// compiler-scheduled instruction can fall between a load and the wait that
// guards its window.  Two banks ping-pong; the current bank's values are
// tied through ("+v") so their consumers order after the wait.
#define DG_STAGE8(c0, c1, c2, c3, c4, c5, c6, c7, n0, n1, n2, n3, n4, n5,  \
                  n6, n7, addr, stride)                                    \
  asm volatile("global_load_dwordx4 %9, %8, off\n\t"                       \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %10, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %11, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %12, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %13, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %14, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %15, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "global_load_dwordx4 %16, %8, off\n\t"                      \
               "v_lshl_add_u64 %8, %8, 0, %17\n\t"                         \
               "s_waitcnt vmcnt(8)"                                        \
               : "+v"(c0), "+v"(c1), "+v"(c2), "+v"(c3), "+v"(c4),         \
                 "+v"(c5), "+v"(c6), "+v"(c7), "+v"(addr), "=&v"(n0),      \
                 "=&v"(n1), "=&v"(n2), "=&v"(n3), "=&v"(n4), "=&v"(n5),    \
                 "=&v"(n6), "=&v"(n7)                                      \
               : "v"(stride))

// prologue: issue one bank, no wait
#define DG_ISSUE8(n0, n1, n2, n3, n4, n5, n6, n7, addr, stride)            \
  asm volatile("global_load_dwordx4 %1, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %2, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %3, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %4, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %5, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %6, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %7, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9\n\t"                          \
               "global_load_dwordx4 %8, %0, off\n\t"                       \
               "v_lshl_add_u64 %0, %0, 0, %9"                              \
               : "+v"(addr), "=&v"(n0), "=&v"(n1), "=&v"(n2), "=&v"(n3),   \
                 "=&v"(n4), "=&v"(n5), "=&v"(n6), "=&v"(n7)                \
               : "v"(stride))

// wait draining everything, tying one bank's consumers behind it
#define DG_DRAIN8(b0, b1, b2, b3, b4, b5, b6, b7)                          \
  asm volatile("s_waitcnt vmcnt(0)"                                        \
               : "+v"(b0), "+v"(b1), "+v"(b2), "+v"(b3), "+v"(b4),         \
                 "+v"(b5), "+v"(b6), "+v"(b7))

template <int QTM>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_pipe2(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  constexpr int RPL = 4;
  extern __shared__ __attribute__((aligned(16))) float smem[];  // [QTM * d]
  int64_t* cbase = (int64_t*)(smem + (size_t)QTM * d);          // [QTM]

  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* col = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];
  const int64_t row0 = list_start + (int64_t)chunk * chunk_rows;
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;

  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  __builtin_assume(d % 4 == 0 && d > 0);

  for (int32_t t0 = 0; t0 < nql; t0 += QTM) {
    const int32_t qt = min(QTM, nql - t0);
    __syncthreads();
    for (int32_t j = 0; j < qt; j++) {
      int32_t q = inv_q[iq0 + t0 + j];
      const float4* src = (const float4*)(queries + (size_t)q * d);
      float4* dst = (float4*)(smem + (size_t)j * d);
      for (int i = threadIdx.x; i < d / 4; i += blockDim.x) dst[i] = src[i];
    }
    for (size_t i = (size_t)qt * d + threadIdx.x; i < (size_t)QTM * d;
         i += blockDim.x)
      smem[i] = 0.f;
    if (threadIdx.x < QTM) {
      int32_t j = threadIdx.x;
      if (j < qt) {
        int32_t q = inv_q[iq0 + t0 + j];
        int32_t rank = inv_rank[iq0 + t0 + j];
        cbase[j] = q_cand_base[q] + qp_off[(int64_t)q * nprobe + rank] -
                   list_start;
      } else {
        cbase[j] = 0;
      }
    }
    __syncthreads();

    for (int32_t rb = wave_id * WAVE * RPL; rb < nrows_pad;
         rb += 4 * WAVE * RPL) {
      const int32_t rr0 = rb + lane * RPL;
      if (rr0 >= nrows_pad) continue;
      float acc[QTM][RPL];
#pragma unroll
      for (int j = 0; j < QTM; j++)
#pragma unroll
        for (int x = 0; x < RPL; x++) acc[j][x] = 0.f;

      // drain previous-tile vmem so the fused counts are exact
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

      const size_t cstride = (size_t)nrows_pad * 4;
      uint64_t addr = (uint64_t)(uintptr_t)(col + rr0);
      const uint64_t stride = cstride;
      dg_f4 a0, a1, a2, a3, a4, a5, a6, a7;
      dg_f4 b0, b1, b2, b3, b4, b5, b6, b7;

      // 8 dims of one bank against all staged queries.  The sched barrier
      // every 4 queries stops the scheduler from hoisting ALL queries' LDS
      // reads at once (measured: 96 live query registers, pushing the
      // allocator into AGPR stashes of the in-flight bank = reads of
      // pending load destinations).
      auto compute8 = [&](const dg_f4& x0, const dg_f4& x1, const dg_f4& x2,
                          const dg_f4& x3, const dg_f4& x4, const dg_f4& x5,
                          const dg_f4& x6, const dg_f4& x7, int32_t ib) {
        const dg_f4* xs[8] = {&x0, &x1, &x2, &x3, &x4, &x5, &x6, &x7};
#pragma unroll
        for (int j = 0; j < QTM; j++) {
          const float* qp = smem + (size_t)j * d + ib;
#pragma unroll
          for (int h = 0; h < 4; h++) {
            const float2 qq = *(const float2*)(qp + 2 * h);
            const dg_f4& ca = *xs[2 * h];
            const dg_f4& cb = *xs[2 * h + 1];
#pragma unroll
            for (int x = 0; x < RPL; x++)
              acc[j][x] += ca[x] * qq.x + cb[x] * qq.y;
          }
          if ((j & 3) == 3) __builtin_amdgcn_sched_barrier(0);
        }
      };

      int32_t base = 0;
      if (d >= 16) {
        DG_ISSUE8(a0, a1, a2, a3, a4, a5, a6, a7, addr, stride);
        for (; base + 16 <= d; base += 16) {
          // sched barriers pin each compute phase between its stage asms:
          // without them the scheduler sinks FMA chunks past the next
          // stage, doubling bank liveness and pushing the allocator into
          // AGPR stashes of in-flight destinations
          DG_STAGE8(a0, a1, a2, a3, a4, a5, a6, a7, b0, b1, b2, b3, b4, b5,
                    b6, b7, addr, stride);
          __builtin_amdgcn_sched_barrier(0);
          compute8(a0, a1, a2, a3, a4, a5, a6, a7, base);
          __builtin_amdgcn_sched_barrier(0);
          DG_STAGE8(b0, b1, b2, b3, b4, b5, b6, b7, a0, a1, a2, a3, a4, a5,
                    a6, a7, addr, stride);
          __builtin_amdgcn_sched_barrier(0);
          compute8(b0, b1, b2, b3, b4, b5, b6, b7, base + 8);
          __builtin_amdgcn_sched_barrier(0);
        }
        DG_DRAIN8(a0, a1, a2, a3, a4, a5, a6, a7);
      }
      // tail (< 16 dims): compiler-managed loads
      const float* colp = col + rr0;
      for (int32_t ib = base; ib < d; ib += 4) {
        float4 c[4];
#pragma unroll
        for (int u = 0; u < 4; u++)
          c[u] = (ib + u < d) ? *(const float4*)(colp +
                                                 (size_t)(ib + u) * nrows_pad)
                              : float4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int j = 0; j < QTM; j++) {
          const float* qp = smem + (size_t)j * d + ib;
#pragma unroll
          for (int u = 0; u < 4 && ib + u < d; u++) {
            const float qv = qp[u];
            acc[j][0] += c[u].x * qv;
            acc[j][1] += c[u].y * qv;
            acc[j][2] += c[u].z * qv;
            acc[j][3] += c[u].w * qv;
          }
        }
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

#pragma unroll
      for (int j = 0; j < QTM; j++) {
        if (j < qt) {
          const int64_t cb = cbase[j] + row0 + rr0;
#pragma unroll
          for (int x = 0; x < RPL; x++) {
            const int32_t rl = rr0 + x;
            if (rl < nrows) {
              const int64_t r = row0 + rl;
              bool pass = true;
              if (bitmap) pass = (bitmap[r >> 5] >> (r & 31)) & 1;
              float key = (metric == 0) ? vnorms[r] - 2.0f * acc[j][x]
                                        : -acc[j][x];
              cand[cb + x] = pass ? pack_cand(key, (uint32_t)r) : kCandEmpty;
            }
          }
        }
      }
    }
  }
}

// ---------- glds scan (v3 experiment, DG_SCAN_VARIANT=7) ----------
// Column data streams HBM -> LDS via global_load_lds into a per-wave
// private ring (each wave owns a 256-row quarter of the chunk, so no
// cross-wave hand-off and no barriers in the steady loop; counted vmcnt
// waits only).  Consumers read columns from LDS (conflict-free b128) and
// accumulate lane-private dots as in v2.  1 block/CU (LDS-bound).
template <int QTM, int DEPTH>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_glds(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  // LDS: [QTM*d] query tile | per-wave rings [4][DEPTH][4*256] | cbase[QTM]
  extern __shared__ __attribute__((aligned(16))) float smem[];
  constexpr int SLOT = 4 * 256;  // 4 dims x 256 rows (f32)
  float* ring_all = smem + (size_t)QTM * d;
  int64_t* cbase = (int64_t*)(ring_all + 4 * DEPTH * SLOT);

  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* col = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];
  const int64_t row0 = list_start + (int64_t)chunk * chunk_rows;
  const int32_t iq0 = inv_offsets[list];
  const int32_t nql = inv_offsets[list + 1] - iq0;

  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  __builtin_assume(d % 4 == 0 && d > 0);
  float* ring = ring_all + wave_id * DEPTH * SLOT;
  const int32_t wrow0 = wave_id * 256;          // this wave's row quarter
  if (wrow0 >= nrows_pad) {
    // idle quarter still participates in the staging barriers below
  }
  const int32_t rr0 = wrow0 + lane * 4;         // this lane's 4 rows
  const int nslots = d / 4;                     // slots per full pass

  for (int32_t t0 = 0; t0 < nql; t0 += QTM) {
    const int32_t qt = min(QTM, nql - t0);
    __syncthreads();
    for (int32_t j = 0; j < qt; j++) {
      int32_t q = inv_q[iq0 + t0 + j];
      const float4* src = (const float4*)(queries + (size_t)q * d);
      float4* dst = (float4*)(smem + (size_t)j * d);
      for (int i = threadIdx.x; i < d / 4; i += blockDim.x) dst[i] = src[i];
    }
    for (size_t i = (size_t)qt * d + threadIdx.x; i < (size_t)QTM * d;
         i += blockDim.x)
      smem[i] = 0.f;
    if (threadIdx.x < QTM) {
      int32_t j = threadIdx.x;
      if (j < qt) {
        int32_t q = inv_q[iq0 + t0 + j];
        int32_t rank = inv_rank[iq0 + t0 + j];
        cbase[j] = q_cand_base[q] + qp_off[(int64_t)q * nprobe + rank] -
                   list_start;
      } else {
        cbase[j] = 0;
      }
    }
    __syncthreads();
    if (wrow0 < nrows_pad) {
      float acc[QTM][4];
#pragma unroll
      for (int j = 0; j < QTM; j++)
#pragma unroll
        for (int x = 0; x < 4; x++) acc[j][x] = 0.f;

      // issue one slot: 4 glds of 1 KiB (dim column slice for this quarter)
      auto issue = [&](int32_t slot_i /* dim block index */) {
        float* ldst = ring + (slot_i % DEPTH) * SLOT;
#pragma unroll
        for (int u = 0; u < 4; u++) {
          const float* gsrc =
              col + (size_t)(slot_i * 4 + u) * nrows_pad + wrow0 + lane * 4;
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) uint32_t*)gsrc,
              (__attribute__((address_space(3))) uint32_t*)(ldst + u * 256),
              16, 0, 0);
        }
      };
      auto consume = [&](int32_t slot_i) {
        const float* sl = ring + (slot_i % DEPTH) * SLOT + lane * 4;
#pragma unroll
        for (int u = 0; u < 4; u++) {
          const float4 c4 = *(const float4*)(sl + u * 256);
#pragma unroll
          for (int j = 0; j < QTM; j++) {
            const float qv = smem[(size_t)j * d + slot_i * 4 + u];
            acc[j][0] += c4.x * qv;
            acc[j][1] += c4.y * qv;
            acc[j][2] += c4.z * qv;
            acc[j][3] += c4.w * qv;
          }
        }
      };
      const int pre = min(DEPTH, nslots);
      for (int s = 0; s < pre; s++) issue(s);
      for (int s = 0; s < nslots; s++) {
        // wait until slot s's 4 glds landed; leave the younger slots'
        // glds in flight: outstanding after wait = 4*(issued - s - 1).
        // asm form with memory clobber so ds_reads can't hoist above it
        // (counted waits only — a vmcnt(0) here would drain the ring).
        const int n_out = 4 * (min(nslots, s + DEPTH) - s - 1);
        switch (n_out) {
          case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
          case 4: asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); break;
          case 8: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
          case 12: asm volatile("s_waitcnt vmcnt(12)" ::: "memory"); break;
          case 16: asm volatile("s_waitcnt vmcnt(16)" ::: "memory"); break;
          case 20: asm volatile("s_waitcnt vmcnt(20)" ::: "memory"); break;
          default: asm volatile("s_waitcnt vmcnt(24)" ::: "memory"); break;
        }
        consume(s);
        if (s + DEPTH < nslots) issue(s + DEPTH);
      }
      (void)pre;
      // emit (guards, not breaks — see v2)
#pragma unroll
      for (int j = 0; j < QTM; j++) {
        if (j < qt) {
          const int64_t cb = cbase[j] + row0 + rr0;
#pragma unroll
          for (int x = 0; x < 4; x++) {
            const int32_t rl = rr0 + x;
            if (rl < nrows) {
              const int64_t r = row0 + rl;
              bool pass = true;
              if (bitmap) pass = (bitmap[r >> 5] >> (r & 31)) & 1;
              float key = (metric == 0) ? vnorms[r] - 2.0f * acc[j][x]
                                        : -acc[j][x];
              cand[cb + x] = pass ? pack_cand(key, (uint32_t)r)
                                  : kCandEmpty;
            }
          }
        }
      }
    }
  }
}

// wrappers: same body, different register budgets.  The (256,1) form lets
// the allocator use 209 VGPRs (2 blocks/CU, LDS would allow 3); the
// (256,3) form caps at 168 VGPRs for 3 waves/SIMD at the cost of possible
// cold spills -- A/B'd via DG_SCAN_VARIANT.
template <int QTM, int RPL>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_col(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  ivf_scan_col_body<QTM, RPL>(units, csr_offsets, chunk_off, chunk_base,
                              tvec, vnorms, queries, d, inv_offsets, inv_q,
                              inv_rank, qp_off, q_cand_base, nprobe, metric,
                              bitmap, chunk_rows, cand);
}

template <int QTM, int RPL>
__global__ void __launch_bounds__(256, 1) k_ivf_scan_col_nt(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  ivf_scan_col_body<QTM, RPL, true>(units, csr_offsets, chunk_off, chunk_base,
                                    tvec, vnorms, queries, d, inv_offsets,
                                    inv_q, inv_rank, qp_off, q_cand_base,
                                    nprobe, metric, bitmap, chunk_rows, cand);
}

template <int QTM, int RPL>
__global__ void __launch_bounds__(256, 3) k_ivf_scan_col_hi(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ tvec, const float* __restrict__ vnorms,
    const float* __restrict__ queries, int32_t d,
    const int32_t* __restrict__ inv_offsets, const int32_t* __restrict__ inv_q,
    const int32_t* __restrict__ inv_rank, const int64_t* __restrict__ qp_off,
    const int64_t* __restrict__ q_cand_base, int32_t nprobe, int metric,
    const uint32_t* __restrict__ bitmap, int32_t chunk_rows,
    uint64_t* __restrict__ cand) {
  ivf_scan_col_body<QTM, RPL>(units, csr_offsets, chunk_off, chunk_base,
                              tvec, vnorms, queries, d, inv_offsets, inv_q,
                              inv_rank, qp_off, q_cand_base, nprobe, metric,
                              bitmap, chunk_rows, cand);
}

// tiled row-major -> column-major chunk transpose (finalize step).
// One block per (list, chunk) unit over ALL chunks; LDS 64x65 tile.
__global__ void k_transpose_chunks(
    const uint32_t* __restrict__ units, const int64_t* __restrict__ csr_offsets,
    const int32_t* __restrict__ chunk_off, const int64_t* __restrict__ chunk_base,
    const float* __restrict__ rowmajor, int32_t d, int32_t chunk_rows,
    float* __restrict__ tvec) {
  __shared__ float tile[64][65];
  const uint32_t list = units[2 * blockIdx.x];
  const uint32_t chunk = units[2 * blockIdx.x + 1];
  const int64_t list_start = csr_offsets[list];
  const int64_t len = csr_offsets[list + 1] - list_start;
  const int32_t nrows =
      (int32_t)min((int64_t)chunk_rows, len - (int64_t)chunk * chunk_rows);
  const int32_t nrows_pad = (nrows + 3) & ~3;
  const float* src =
      rowmajor + (list_start + (int64_t)chunk * chunk_rows) * d;
  float* dst = tvec + chunk_base[chunk_off[list] + (int32_t)chunk];

  const int tx = threadIdx.x % 64;  // dim within tile on read, row on write
  const int ty = threadIdx.x / 64;  // 4 groups
  for (int32_t r0 = 0; r0 < nrows; r0 += 64) {
    for (int32_t i0 = 0; i0 < d; i0 += 64) {
      __syncthreads();
      for (int rr = ty; rr < 64; rr += 4) {
        int32_t r = r0 + rr;
        tile[rr][tx] = (r < nrows && i0 + tx < d)
                           ? src[(size_t)r * d + i0 + tx]
                           : 0.f;
      }
      __syncthreads();
      for (int ii = ty; ii < 64; ii += 4) {
        int32_t i = i0 + ii;
        int32_t r = r0 + tx;
        if (i < d && r < nrows_pad)
          dst[(size_t)i * nrows_pad + r] = tile[tx][ii];
      }
    }
  }
}

// ---------- hand MFMA f32 dots GEMM ----------
// C[M][N] = X[M][d] . Y[N][d]^T, all row-major — the coarse
// query x centroid assignment GEMM the north star names as THE MFMA target
// (SURVEY.md §8a row a4), also used for the Flat exact-scan dots.
// gfx950 f32-input MFMA v_mfma_f32_32x32x2_f32: exact f32 (a k-ordered fmaf
// chain, cdna_hip_programming.md §3), 64 cyc issue = dependent latency, so
// one accumulator per wave reaches the 157 TF f32 rate.  Block = 4 waves =
// 64x64 tile (each wave one 32x32 accumulator, 16 AGPRs), BK=16 double-
// buffered LDS (2 x 2 x 64x16 f32 = 16 KB), A/B operands read from LDS as
// b64 (two k-steps per read).
__global__ void __launch_bounds__(256, 2) k_dots_mfma(
    const float* __restrict__ X, const float* __restrict__ Y, int64_t M,
    int64_t N, int32_t K, float* __restrict__ C, int64_t ldc) {
  // 128x128 block, 4 waves as 2x2 of 64x64, each wave 2x2 accumulators of
  // 32x32 => 64 MFMA per wave between barriers (the guide's untuned
  // 128x128x32 f32 shape reaches 122 TF; a 64x64/BK16 first cut measured
  // 27 TF — barrier overhead per 8 MFMAs dominated).
  constexpr int BM = 128, BN = 128, BK = 32;
  // BK+4 stride: rows are 144 B = 16 B-aligned, so staging writes are
  // ds_write_b128 (the round-1 BK+1 odd stride forced 32 SCALAR b32
  // writes per thread per stage — the staging cost, not the MFMAs, capped
  // that kernel at 57 TF).  Column reads pay a 2-way bank conflict
  // (rows r and r+16 share banks at stride 36) — measured cheaper than
  // the scalar-write staging.
  __shared__ float lx[2][BM][BK + 4];
  __shared__ float ly[2][BN][BK + 4];
  using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

  const int64_t m0 = (int64_t)blockIdx.y * BM;
  const int64_t n0 = (int64_t)blockIdx.x * BN;
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wm = (wave_id >> 1) * 64;  // wave row offset in block tile
  const int wn = (wave_id & 1) * 64;   // wave col offset

  // staging: 128 rows x 32 k per operand; thread t loads 16 consecutive k
  // of one row (4 x float4 when fully in range)
  const int sr = threadIdx.x / 2;         // staged row 0..127
  const int sk = (threadIdx.x % 2) * 16;  // staged k 0 or 16

  f32x16 acc[2][2] = {};
  auto stage = [&](int buf, int32_t k0) {
    const int64_t xm = m0 + sr;
    const int64_t yn = n0 + sr;
    if (xm < M && yn < N && k0 + sk + 16 <= K) {
#pragma unroll
      for (int t = 0; t < 4; t++) {
        const float4 vx = *(const float4*)&X[xm * K + k0 + sk + 4 * t];
        const float4 vy = *(const float4*)&Y[yn * K + k0 + sk + 4 * t];
        *(float4*)&lx[buf][sr][sk + 4 * t] = vx;
        *(float4*)&ly[buf][sr][sk + 4 * t] = vy;
      }
    } else {  // edge tile: element-wise with zero padding
#pragma unroll
      for (int t = 0; t < 16; t++) {
        const int32_t k = k0 + sk + t;
        lx[buf][sr][sk + t] = (xm < M && k < K) ? X[xm * K + k] : 0.f;
        ly[buf][sr][sk + t] = (yn < N && k < K) ? Y[yn * K + k] : 0.f;
      }
    }
  };
  auto compute = [&](int buf) {
    const int al = lane & 31;   // row/col within a 32-wide fragment
    const int kh = (lane >> 5); // k half within each k-pair
#pragma unroll
    for (int kk = 0; kk < BK; kk += 2) {
      float a[2], b[2];
#pragma unroll
      for (int i = 0; i < 2; i++) {
        a[i] = lx[buf][wm + 32 * i + al][kk + kh];
        b[i] = ly[buf][wn + 32 * i + al][kk + kh];
      }
#pragma unroll
      for (int mi = 0; mi < 2; mi++)
#pragma unroll
        for (int ni = 0; ni < 2; ni++)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  stage(0, 0);
  int buf = 0;
  for (int32_t k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    if (k0 + BK < K) stage(buf ^ 1, k0 + BK);
    compute(buf);
    buf ^= 1;
    __syncthreads();
  }
  // C/D layout for 32x32 f32 MFMA: col = lane&31,
  // row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)   (cdna_hip_programming.md §3)
#pragma unroll
  for (int mi = 0; mi < 2; mi++)
#pragma unroll
    for (int ni = 0; ni < 2; ni++)
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        const int64_t cm = m0 + wm + 32 * mi + row;
        const int64_t cn = n0 + wn + 32 * ni + (lane & 31);
        if (cm < M && cn < N) C[cm * ldc + cn] = acc[mi][ni][reg];
      }
}

// ---------- k-means / finalize helpers ----------
__global__ void k_hist_assign(const int32_t* __restrict__ assign, int64_t n,
                              int32_t* __restrict__ counts) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) atomicAdd(&counts[assign[i]], 1);
}

__global__ void k_init_cursors(const int64_t* __restrict__ offsets, int32_t n,
                               int32_t* __restrict__ cursors) {
  int32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) cursors[i] = (int32_t)offsets[i];
}

__global__ void k_scatter_perm(const int32_t* __restrict__ assign, int64_t n,
                               int32_t* __restrict__ cursors,
                               uint32_t* __restrict__ perm) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  perm[i] = (uint32_t)atomicAdd(&cursors[assign[i]], 1);
}

__global__ void k_gather_rows(const float* __restrict__ src,
                              const uint32_t* __restrict__ perm, int64_t n,
                              int32_t d, float* __restrict__ dst) {
  // wave per row: dst[perm[i]] = src[i]
  int64_t i = (int64_t)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  if (i >= n) return;
  const float4* s = (const float4*)(src + (size_t)i * d);
  float4* t = (float4*)(dst + (size_t)perm[i] * d);
  for (int j = lane; j < d / 4; j += WAVE) t[j] = s[j];
}

__global__ void k_gather_ids(const int64_t* __restrict__ src,
                             const uint32_t* __restrict__ perm, int64_t n,
                             int64_t* __restrict__ dst) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[perm[i]] = src[i];
}

__global__ void k_cluster_means(const float* __restrict__ grouped,
                                const int64_t* __restrict__ offsets,
                                int32_t d, float* __restrict__ centroids) {
  // block per cluster; thread t owns dims t, t+256, ...
  // acc sized for d <= 8192 at blockDim 256 (the create-time ceiling;
  // acc[8] overflowed for IVF-PQ trains with d > 2048)
  int32_t c = blockIdx.x;
  int64_t s = offsets[c], e = offsets[c + 1];
  if (e <= s) return;  // empty: host handles split
  double acc[32] = {0};
  int nd = (d + blockDim.x - 1) / blockDim.x;
  if (nd > 32) return;  // host enforces d <= 8192
  for (int64_t r = s; r < e; r++) {
    const float* v = grouped + (size_t)r * d;
    for (int i = 0; i < nd; i++) {
      int dim = threadIdx.x + i * blockDim.x;
      if (dim < d) acc[i] += v[dim];
    }
  }
  double inv = 1.0 / (double)(e - s);
  for (int i = 0; i < nd; i++) {
    int dim = threadIdx.x + i * blockDim.x;
    if (dim < d) centroids[(size_t)c * d + dim] = (float)(acc[i] * inv);
  }
}

// ---------- exclusive scans (small n: two-level) ----------
__global__ void k_scan_block_i32(const int32_t* in, int32_t n, int64_t* out,
                                 int64_t* block_sums) {
  __shared__ int64_t lds[256];
  int32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  int64_t v = (i < n) ? in[i] : 0;
  lds[threadIdx.x] = v;
  __syncthreads();
  // inclusive scan in LDS
  for (int off = 1; off < 256; off *= 2) {
    int64_t t = (threadIdx.x >= off) ? lds[threadIdx.x - off] : 0;
    __syncthreads();
    lds[threadIdx.x] += t;
    __syncthreads();
  }
  if (i <= n) out[i] = lds[threadIdx.x] - v;  // exclusive
  if (threadIdx.x == 255) block_sums[blockIdx.x] = lds[255];
}

__global__ void k_scan_add_offsets(int64_t* out, int32_t n,
                                   const int64_t* block_sums, int32_t nblocks) {
  // single block: serial scan of block sums (nblocks small), then add
  __shared__ int64_t bs[1024];
  if (threadIdx.x == 0) {
    int64_t run = 0;
    for (int b = 0; b < nblocks; b++) {
      bs[b] = run;
      run += block_sums[b];
    }
    bs[nblocks] = run;
  }
  __syncthreads();
  for (int32_t i = threadIdx.x; i <= n; i += blockDim.x)
    out[i] += bs[min(i / 256, nblocks)];
}

// i64 variant (for q_total -> q_cand_base)
__global__ void k_scan_block_i64(const int64_t* in, int64_t n, int64_t* out,
                                 int64_t* block_sums) {
  __shared__ int64_t lds[256];
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t v = (i < n) ? in[i] : 0;
  lds[threadIdx.x] = v;
  __syncthreads();
  for (int off = 1; off < 256; off *= 2) {
    int64_t t = (threadIdx.x >= off) ? lds[threadIdx.x - off] : 0;
    __syncthreads();
    lds[threadIdx.x] += t;
    __syncthreads();
  }
  if (i <= n) out[i] = lds[threadIdx.x] - v;
  if (threadIdx.x == 255) block_sums[blockIdx.x] = lds[255];
}

// ---------- launchers ----------
namespace dgk {

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

void iota_i32(hipStream_t s, int32_t* p, int64_t n, int32_t v) {
  if (n) hipLaunchKernelGGL(k_iota_i32, dim3(ceil_div(n, 256)), dim3(256), 0, s,
                            p, n, v);
}

void row_norms(hipStream_t s, const float* x, int64_t n, int32_t d,
               float* out) {
  if (!n) return;
  int wpb = 4;
  hipLaunchKernelGGL(k_row_norms, dim3(ceil_div(n, wpb)), dim3(wpb * WAVE), 0,
                     s, x, n, d, out);
}

void normalize_rows(hipStream_t s, float* x, int64_t n, int32_t d) {
  if (!n) return;
  int wpb = 4;
  hipLaunchKernelGGL(k_normalize_rows, dim3(ceil_div(n, wpb)),
                     dim3(wpb * WAVE), 0, s, x, n, d);
}

void argmin_rows(hipStream_t s, const float* dots, const float* cnorms,
                 int64_t n, int32_t nlist, int metric, int32_t* out) {
  if (!n) return;
  int wpb = 4;
  hipLaunchKernelGGL(k_argmin_rows, dim3(ceil_div(n, wpb)), dim3(wpb * WAVE),
                     0, s, dots, cnorms, n, nlist, metric, out);
}

void build_pass_bitmap(hipStream_t s, const int64_t* ids, int64_t n,
                       const dg_dev_filter* f, uint32_t* bitmap) {
  int64_t nwords = ceil_div(n, 32);
  hipLaunchKernelGGL(k_build_pass_bitmap, dim3(ceil_div(nwords, 256)),
                     dim3(256), 0, s, ids, n, *f, bitmap);
}

// choose block size by k: 2*T*k*8 <= 128 KB dynamic LDS (measured-good on
// gfx950).  Blocks shrink below a wave for the rare huge-k calls — correct,
// slow, and documented; host validates k <= 2048.
static inline int select_threads(int32_t k) {
  int T = 256;  // largest block whose per-thread lists fit 128 KB LDS
  while (T > 2 && 2ull * T * k * 8 > (128u << 10)) T >>= 1;
  return T;
}

int select_dense_threads(int32_t k) { return select_threads(k); }

void select_dense(hipStream_t s, const float* scores, const float* cnorms,
                  int64_t rows, int64_t cols, int64_t ld, int32_t nseg,
                  int32_t k, int mode, const uint32_t* bitmap,
                  int64_t col_base, uint64_t* out, int64_t out_stride,
                  int64_t out_offset) {
  int T = select_threads(k);
  size_t lds = 2ull * T * k * 8;
  int64_t seg_w = (cols + nseg - 1) / nseg;
  hipLaunchKernelGGL(k_select_dense, dim3((uint32_t)rows, (uint32_t)nseg),
                     dim3(T), lds, s, scores, cnorms, rows, cols, ld, seg_w,
                     k, mode, bitmap, col_base, out, out_stride, out_offset);
}

void select_u64(hipStream_t s, const uint64_t* cand, const int64_t* base,
                const int64_t* total, int64_t nq, int32_t k, uint64_t* out,
                int64_t out_stride) {
  int T = select_threads(k);
  size_t lds = 2ull * T * k * 8;
  hipLaunchKernelGGL(k_select_u64, dim3((uint32_t)nq), dim3(T), lds, s, cand,
                     base, total, nq, k, out, out_stride);
}

void emit_results(hipStream_t s, const uint64_t* topk,
                  const int64_t* ids_lookup, const float* qnorms, int64_t nq,
                  int32_t k, int metric, int add_qnorm, float* out_dist,
                  int64_t* out_ids) {
  hipLaunchKernelGGL(k_emit, dim3(ceil_div(nq * k, 256)), dim3(256), 0, s,
                     topk, ids_lookup, qnorms, nq, k, metric, add_qnorm,
                     out_dist, out_ids);
}

void probes_all(hipStream_t s, int64_t nq, int32_t nprobe,
                const uint8_t* mask, int32_t* probes) {
  hipLaunchKernelGGL(k_probes_all, dim3(ceil_div(nq * nprobe, 256)),
                     dim3(256), 0, s, nq, nprobe, mask, probes);
}

void probe_unpack(hipStream_t s, const uint64_t* topk, int64_t nq,
                  int32_t nprobe, const uint8_t* mask, int32_t* probes) {
  hipLaunchKernelGGL(k_probe_unpack, dim3(ceil_div(nq * nprobe, 256)),
                     dim3(256), 0, s, topk, nq, nprobe, mask, probes);
}

void hist_probes(hipStream_t s, const int32_t* probes, int64_t nq,
                 int32_t nprobe, int32_t nlist, int32_t* counts) {
  hipLaunchKernelGGL(k_hist_probes, dim3(ceil_div(nq * nprobe, 256)),
                     dim3(256), 0, s, probes, nq * nprobe, counts);
}

void scatter_probes(hipStream_t s, const int32_t* probes, int64_t nq,
                    int32_t nprobe, const int32_t* /*inv_offsets*/,
                    int32_t* cursors, int32_t* inv_q, int32_t* inv_rank) {
  hipLaunchKernelGGL(k_scatter_probes, dim3(ceil_div(nq * nprobe, 256)),
                     dim3(256), 0, s, probes, nq, nprobe, cursors, inv_q,
                     inv_rank);
}

void cand_offsets(hipStream_t s, const int32_t* probes, int64_t nq,
                  int32_t nprobe, const int64_t* csr_offsets, int64_t* qp_off,
                  int64_t* q_total) {
  hipLaunchKernelGGL(k_cand_offsets, dim3(ceil_div(nq, 128)), dim3(128), 0, s,
                     probes, nq, nprobe, csr_offsets, qp_off, q_total);
}

void alg_bytes(hipStream_t s, const int32_t* inv_counts,
               const int64_t* csr_offsets, int32_t nlist, int64_t row_bytes,
               int64_t* out) {
  hipLaunchKernelGGL(k_alg_bytes, dim3(16), dim3(256), 0, s, inv_counts,
                     csr_offsets, nlist, row_bytes, out);
}

void fill_unit_counts(hipStream_t s, const int32_t* inv_counts, int32_t nlist,
                      const int64_t* csr_offsets, int32_t chunk_rows,
                      int32_t* unit_counts) {
  hipLaunchKernelGGL(k_unit_counts, dim3(ceil_div(nlist, 128)), dim3(128), 0,
                     s, inv_counts, nlist, csr_offsets, chunk_rows,
                     unit_counts);
}

void fill_units(hipStream_t s, const int32_t* unit_offsets,
                const int32_t* unit_counts, int32_t nlist,
                const int64_t* /*csr_offsets*/, int32_t /*chunk_rows*/,
                uint32_t* units, int32_t /*total*/) {
  hipLaunchKernelGGL(k_fill_units, dim3(ceil_div(nlist, 128)), dim3(128), 0,
                     s, unit_offsets, unit_counts, nlist, units);
}

void ivf_scan_col(hipStream_t s, const uint32_t* units, int32_t n_units,
                  const int64_t* csr_offsets, const int32_t* chunk_off,
                  const int64_t* chunk_base, const float* tvec,
                  const float* vnorms, const float* queries, int32_t d,
                  const int32_t* inv_offsets, const int32_t* inv_q,
                  const int32_t* inv_rank, const int64_t* qp_off,
                  const int64_t* q_cand_base, int32_t nprobe, int metric,
                  const uint32_t* bitmap, int32_t chunk_rows,
                  uint64_t* cand, int32_t mean_probes) {
  if (!n_units) return;
  static int variant = []() {
    const char* e = getenv("DG_SCAN_VARIANT");
    return e ? atoi(e) : 0;
  }();
#define DG_SCAN_LAUNCH(QTM, RPL)                                            \
  do {                                                                      \
    size_t lds = (size_t)(QTM) * d * 4 + (QTM) * 8;                         \
    hipLaunchKernelGGL((k_ivf_scan_col<(QTM), (RPL)>),                      \
                       dim3((uint32_t)n_units), dim3(256), lds, s, units,   \
                       csr_offsets, chunk_off, chunk_base, tvec, vnorms,    \
                       queries, d, inv_offsets, inv_q, inv_rank, qp_off,    \
                       q_cand_base, nprobe, metric, bitmap, chunk_rows,     \
                       cand);                                               \
  } while (0)
  switch (variant) {
    case 1: DG_SCAN_LAUNCH(8, 4); break;
    case 2: DG_SCAN_LAUNCH(16, 2); break;
    case 3: DG_SCAN_LAUNCH(8, 8); break;
    case 4: DG_SCAN_LAUNCH(4, 4); break;
    case 5: DG_SCAN_LAUNCH(12, 4); break;
    case 8: {
      constexpr int QTM = 16;
      size_t lds = (size_t)QTM * d * 4 + QTM * 8;
      hipLaunchKernelGGL((k_ivf_scan_col_nt<QTM, 4>),
                         dim3((uint32_t)n_units), dim3(256), lds, s, units,
                         csr_offsets, chunk_off, chunk_base, tvec, vnorms,
                         queries, d, inv_offsets, inv_q, inv_rank, qp_off,
                         q_cand_base, nprobe, metric, bitmap, chunk_rows,
                         cand);
      break;
    }
    case 7: {
      constexpr int QTM = 16, DEPTH = 6;
      size_t lds = (size_t)QTM * d * 4 + 4 * DEPTH * 4 * 256 * 4 + QTM * 8;
      hipLaunchKernelGGL((k_ivf_scan_glds<QTM, DEPTH>),
                         dim3((uint32_t)n_units), dim3(256), lds, s, units,
                         csr_offsets, chunk_off, chunk_base, tvec, vnorms,
                         queries, d, inv_offsets, inv_q, inv_rank, qp_off,
                         q_cand_base, nprobe, metric, bitmap, chunk_rows,
                         cand);
      break;
    }
    case 6: {
      constexpr int QTM = 16;
      size_t lds = (size_t)QTM * d * 4 + QTM * 8;
      hipLaunchKernelGGL((k_ivf_scan_col_hi<QTM, 4>),
                         dim3((uint32_t)n_units), dim3(256), lds, s, units,
                         csr_offsets, chunk_off, chunk_base, tvec, vnorms,
                         queries, d, inv_offsets, inv_q, inv_rank, qp_off,
                         q_cand_base, nprobe, metric, bitmap, chunk_rows,
                         cand);
      break;
    }
    case 0:
    case 12: {  // DEFAULT (round 2): asm-load pipeline with full-drain
      // waits — the counted-wait variants (10/13/14) are faster on paper
      // but LLVM's allocator cannot be made to respect in-flight asm-load
      // destinations (verified by static hazard scan + GPU faults); the
      // drain form is airtight and measured fastest of the correct set.
      // QTM shrinks for large d so the query tile fits the 160 KB LDS.
#define DG_PIPE_LAUNCH(QTM)                                                 \
  do {                                                                      \
    size_t lds = (size_t)(QTM) * d * 4 + (QTM) * 8;                         \
    hipLaunchKernelGGL((k_ivf_scan_pipe<(QTM), true>),                      \
                       dim3((uint32_t)n_units), dim3(256), lds, s, units,   \
                       csr_offsets, chunk_off, chunk_base, tvec, vnorms,    \
                       queries, d, inv_offsets, inv_q, inv_rank, qp_off,    \
                       q_cand_base, nprobe, metric, bitmap, chunk_rows,     \
                       cand);                                               \
  } while (0)
      (void)mean_probes;  // dense-batch 20/24-query tiles fail the
                          // in-flight hazard scan (allocator stashes
                          // pending asm loads at >=250 VGPRs) — QTM=16
                          // is the largest clean tile
      if (d <= 2304)
        DG_PIPE_LAUNCH(16);
      else if (d <= 4608)
        DG_PIPE_LAUNCH(8);
      else
        DG_PIPE_LAUNCH(4);  // d <= 8192 (create-time cap)
#undef DG_PIPE_LAUNCH
      break;
    }
    // (case 19 pipe4 removed: 2x8-bank wait/issue/compute order also
    // fails the hazard scan — 8 pending loads across a full compute push
    // the allocator into stashing in-flight destinations)
    case 15: {  // round-1 columnar pipeline (previous default)
      DG_SCAN_LAUNCH(16, 4);
      break;
    }
// (pipe2/pipe3 launch cases removed: every counted-wait or batched-issue
// schedule fails the static in-flight-register hazard scan — kernels kept
// above as the documented record of the attempt)
    case 9: {  // (12,4) capped to 3 waves/SIMD (uncapped allocates 177)
      constexpr int QTM = 12;
      size_t lds = (size_t)QTM * d * 4 + QTM * 8;
      hipLaunchKernelGGL((k_ivf_scan_col_hi<QTM, 4>),
                         dim3((uint32_t)n_units), dim3(256), lds, s, units,
                         csr_offsets, chunk_off, chunk_base, tvec, vnorms,
                         queries, d, inv_offsets, inv_q, inv_rank, qp_off,
                         q_cand_base, nprobe, metric, bitmap, chunk_rows,
                         cand);
      break;
    }
    default: DG_SCAN_LAUNCH(16, 4); break;
  }
#undef DG_SCAN_LAUNCH
}

void range_emit(hipStream_t s, const uint64_t* packed, const int64_t* lims,
                const int64_t* ids_lookup, const float* qnorms, int64_t nq,
                int metric, int add_qnorm, float* out_dist,
                int64_t* out_ids) {
  hipLaunchKernelGGL(k_range_emit, dim3((uint32_t)nq), dim3(256), 0, s,
                     packed, lims, ids_lookup, qnorms, nq, metric, add_qnorm,
                     out_dist, out_ids);
}

void count_below(hipStream_t s, const uint64_t* cand, const int64_t* base,
                 const int64_t* total, const uint64_t* thr, int64_t nq,
                 int64_t* counts) {
  hipLaunchKernelGGL(k_count_below, dim3((uint32_t)nq), dim3(256), 0, s,
                     cand, base, total, thr, nq, counts);
}

void compact_below(hipStream_t s, const uint64_t* cand, const int64_t* base,
                   const int64_t* total, const uint64_t* thr,
                   const int64_t* out_off, int64_t nq, int64_t* cursors,
                   uint64_t* out) {
  hipLaunchKernelGGL(k_compact_below, dim3((uint32_t)nq), dim3(256), 0, s,
                     cand, base, total, thr, out_off, nq, cursors, out);
}

void count_below_dense(hipStream_t s, const float* scores,
                       const float* cnorms, int64_t rows, int64_t cols,
                       int mode, const uint32_t* bitmap, int64_t col_base,
                       const uint64_t* thr, int64_t* counts) {
  hipLaunchKernelGGL(k_count_below_dense, dim3((uint32_t)rows), dim3(256), 0,
                     s, scores, cnorms, rows, cols, mode, bitmap, col_base,
                     thr, counts);
}

void compact_below_dense(hipStream_t s, const float* scores,
                         const float* cnorms, int64_t rows, int64_t cols,
                         int mode, const uint32_t* bitmap, int64_t col_base,
                         const uint64_t* thr, const int64_t* out_off,
                         int64_t* cursors, uint64_t* out) {
  hipLaunchKernelGGL(k_compact_below_dense, dim3((uint32_t)rows), dim3(256),
                     0, s, scores, cnorms, rows, cols, mode, bitmap,
                     col_base, thr, out_off, cursors, out);
}

void residual(hipStream_t s, const float* x, const int32_t* assign,
              const float* centroids, int64_t n, int32_t d, float* out) {
  if (n) hipLaunchKernelGGL(k_residual, dim3(ceil_div(n, 4)), dim3(4 * WAVE),
                            0, s, x, assign, centroids, n, d, out);
}

void set_code(hipStream_t s, const int32_t* amin, int64_t n, int32_t m,
              int32_t M, uint8_t* codes) {
  if (n) hipLaunchKernelGGL(k_set_code, dim3(ceil_div(n, 256)), dim3(256), 0,
                            s, amin, n, m, M, codes);
}

void gather_codes(hipStream_t s, const uint8_t* src, const uint32_t* perm,
                  int64_t n, int32_t M, uint8_t* dst) {
  if (n) hipLaunchKernelGGL(k_gather_codes, dim3(ceil_div(n, 8)),
                            dim3(8 * 32), 0, s, src, perm, n, M, dst);
}

void build_S(hipStream_t s, const float* centroids, const float* codebooks,
             int32_t nlist, int32_t M, int32_t dsub, int32_t d, __half* S) {
  int64_t total = (int64_t)nlist * M * 256;
  hipLaunchKernelGGL(k_build_S, dim3(ceil_div(total, 256)), dim3(256), 0, s,
                     centroids, codebooks, nlist, M, dsub, d, S);
}

void f32_to_f16(hipStream_t s, const float* in, int64_t n, __half* out) {
  if (n) hipLaunchKernelGGL(k_f32_to_f16, dim3(ceil_div(n, 256)), dim3(256),
                            0, s, in, n, out);
}

void ivfpq_scan(hipStream_t s, const uint32_t* units, int32_t n_units,
                const int64_t* csr_offsets, const uint8_t* csr_codes,
                const __half* S, const __half* T, const float* coarse_dots,
                int32_t nlist, int32_t M, const int32_t* inv_offsets,
                const int32_t* inv_q, const int32_t* inv_rank,
                const int64_t* qp_off, const int64_t* q_cand_base,
                int32_t nprobe, int metric, const uint32_t* bitmap,
                int32_t chunk_rows, uint64_t* cand) {
  if (!n_units) return;
  // T-table (L2/L3) traffic per chunk scales as 1/TILE: every staged tile
  // walks each probing query's 48 KB T window once, so the largest tile
  // whose codes + S_l fit LDS wins unless occupancy dominates (A/B'd;
  // DG_PQ_RPV forces a tile for experiments).  TILE = RPV * 64 rows.
  const char* e = getenv("DG_PQ_RPV");
  int rpv = e ? atoi(e) : 4;  // 256-row tiles + 2 blocks/CU measured best
  if (rpv != 4 && rpv != 6 && rpv != 8 && rpv != 16 && rpv != 17) rpv = 4;
  size_t lds = (size_t)(rpv == 6 ? 4 : (rpv == 17 ? 16 : rpv)) * 64 * M +
               (size_t)M * 256 * 2;  // codes + f16 S_l
  if (rpv == 17) {  // cooperative: 4 waves share each query's T window
    hipLaunchKernelGGL((k_ivfpq_scan<16, 1, true>), dim3((uint32_t)n_units),
                       dim3(256), lds, s, units, csr_offsets, csr_codes, S,
                       T, coarse_dots, nlist, M, inv_offsets, inv_q,
                       inv_rank, qp_off, q_cand_base, nprobe, metric, bitmap,
                       chunk_rows, cand);
    return;
  }
  if (rpv == 6)  // round-1 form: VGPRs capped at 80 by launch_bounds(256,6)
    hipLaunchKernelGGL((k_ivfpq_scan<4, 6>), dim3((uint32_t)n_units),
                       dim3(256), lds, s, units, csr_offsets, csr_codes, S,
                       T, coarse_dots, nlist, M, inv_offsets, inv_q,
                       inv_rank, qp_off, q_cand_base, nprobe, metric, bitmap,
                       chunk_rows, cand);
  else if (rpv == 16)
    hipLaunchKernelGGL((k_ivfpq_scan<16, 1>), dim3((uint32_t)n_units), dim3(256),
                       lds, s, units, csr_offsets, csr_codes, S, T,
                       coarse_dots, nlist, M, inv_offsets, inv_q, inv_rank,
                       qp_off, q_cand_base, nprobe, metric, bitmap,
                       chunk_rows, cand);
  else if (rpv == 8)
    hipLaunchKernelGGL((k_ivfpq_scan<8, 1>), dim3((uint32_t)n_units), dim3(256),
                       lds, s, units, csr_offsets, csr_codes, S, T,
                       coarse_dots, nlist, M, inv_offsets, inv_q, inv_rank,
                       qp_off, q_cand_base, nprobe, metric, bitmap,
                       chunk_rows, cand);
  else  // default: 256-row tiles with the register cap lifted (2 blocks/CU
        // by LDS anyway; 80-VGPR cap limited outstanding T gathers)
    hipLaunchKernelGGL((k_ivfpq_scan<4, 2>), dim3((uint32_t)n_units),
                       dim3(256), lds, s, units, csr_offsets, csr_codes, S,
                       T, coarse_dots, nlist, M, inv_offsets, inv_q,
                       inv_rank, qp_off, q_cand_base, nprobe, metric, bitmap,
                       chunk_rows, cand);
}

void transpose_chunks(hipStream_t s, const uint32_t* units, int32_t n_units,
                      const int64_t* csr_offsets, const int32_t* chunk_off,
                      const int64_t* chunk_base, const float* rowmajor,
                      int32_t d, int32_t chunk_rows, float* tvec) {
  if (!n_units) return;
  hipLaunchKernelGGL(k_transpose_chunks, dim3((uint32_t)n_units), dim3(256),
                     0, s, units, csr_offsets, chunk_off, chunk_base,
                     rowmajor, d, chunk_rows, tvec);
}

void dots_mfma(hipStream_t s, const float* X, int64_t M, const float* Y,
               int64_t N, int32_t K, float* C, int64_t ldc) {
  dim3 grid((uint32_t)((N + 127) / 128), (uint32_t)((M + 127) / 128));
  hipLaunchKernelGGL(k_dots_mfma, grid, dim3(256), 0, s, X, Y, M, N, K, C,
                     ldc);
}

void hist_assign(hipStream_t s, const int32_t* assign, int64_t n,
                 int32_t nlist, int32_t* counts) {
  hipLaunchKernelGGL(k_hist_assign, dim3(ceil_div(n, 256)), dim3(256), 0, s,
                     assign, n, counts);
}

void init_cursors(hipStream_t s, const int64_t* offsets, int32_t n,
                  int32_t* cursors) {
  hipLaunchKernelGGL(k_init_cursors, dim3(ceil_div(n, 128)), dim3(128), 0, s,
                     offsets, n, cursors);
}

void scatter_perm(hipStream_t s, const int32_t* assign, int64_t n,
                  const int64_t* /*offsets*/, int32_t* cursors,
                  uint32_t* perm) {
  hipLaunchKernelGGL(k_scatter_perm, dim3(ceil_div(n, 256)), dim3(256), 0, s,
                     assign, n, cursors, perm);
}

void gather_rows(hipStream_t s, const float* src, const uint32_t* perm,
                 int64_t n, int32_t d, float* dst) {
  if (!n) return;
  int wpb = 4;
  hipLaunchKernelGGL(k_gather_rows, dim3(ceil_div(n, wpb)), dim3(wpb * WAVE),
                     0, s, src, perm, n, d, dst);
}

void gather_ids(hipStream_t s, const int64_t* src, const uint32_t* perm,
                int64_t n, int64_t* dst) {
  hipLaunchKernelGGL(k_gather_ids, dim3(ceil_div(n, 256)), dim3(256), 0, s,
                     src, perm, n, dst);
}

void cluster_means(hipStream_t s, const float* grouped, const int64_t* offsets,
                   int32_t nlist, int32_t d, float* centroids) {
  hipLaunchKernelGGL(k_cluster_means, dim3(nlist), dim3(256), 0, s, grouped,
                     offsets, d, centroids);
}

void excl_scan_i32_to_i64(hipStream_t s, const int32_t* in, int32_t n,
                          int64_t* out, int64_t* bs) {
  int nblocks = (int)ceil_div(n + 1, 256);  // thread i==n writes the total
  hipLaunchKernelGGL(k_scan_block_i32, dim3(nblocks), dim3(256), 0, s, in, n,
                     out, bs);
  hipLaunchKernelGGL(k_scan_add_offsets, dim3(1), dim3(256), 0, s, out, n, bs,
                     nblocks);
}

void excl_scan_i64(hipStream_t s, const int64_t* in, int64_t n, int64_t* out,
                   int64_t* bs) {
  int nblocks = (int)ceil_div(n + 1, 256);
  hipLaunchKernelGGL(k_scan_block_i64, dim3(nblocks), dim3(256), 0, s, in, n,
                     out, bs);
  hipLaunchKernelGGL(k_scan_add_offsets, dim3(1), dim3(256), 0, s, out,
                     (int32_t)n, bs, nblocks);
}

void gather_rows_by_index(hipStream_t s, const float* src, const int64_t* idx,
                          int64_t n, int32_t d, float* dst) {
  if (!n) return;
  hipLaunchKernelGGL(k_gather_rows_by_index, dim3(ceil_div(n, 4)),
                     dim3(4 * WAVE), 0, s, src, idx, n, d, dst);
}

void fill_base_total(hipStream_t s, int64_t nq, int64_t len, int64_t* base,
                     int64_t* total) {
  hipLaunchKernelGGL(k_fill_base_total, dim3(ceil_div(nq, 256)), dim3(256), 0,
                     s, nq, len, base, total);
}

void tombstone(hipStream_t s, const int64_t* pos, int64_t n, int64_t* ids) {
  if (n) hipLaunchKernelGGL(k_tombstone, dim3(ceil_div(n, 256)), dim3(256), 0,
                            s, pos, n, ids);
}

}  // namespace dgk
