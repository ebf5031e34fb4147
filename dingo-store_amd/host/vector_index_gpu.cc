// vector_index_gpu.cc — GPU-backed implementations of the VectorIndex
// plugin mirror.  Thin shim over the C-ABI: marshals the plain-struct proto
// re-declarations to packed arrays (restating ExtractVectorValue,
// src/vector/vector_index_utils.cc:564-609) and shapes results exactly like
// FillSearchResult (:612-655): L2 raw squared distance passes through,
// IP/cosine reported as 1.0f - score (:634), labels < 0 skipped.
#include "vector_index_gpu.h"

#include <cmath>
#include <cstdio>
#include <cstring>

#include <algorithm>
#include <queue>
#include <unordered_set>

namespace dingogpu {

ConcreteFilterFunctor::ConcreteFilterFunctor(const std::vector<int64_t>& ids,
                                             bool is_negation)
    : sorted_(ids), negation_(is_negation) {
  std::sort(sorted_.begin(), sorted_.end());
  sorted_.erase(std::unique(sorted_.begin(), sorted_.end()), sorted_.end());
}

bool ConcreteFilterFunctor::Check(int64_t id) {
  bool found = std::binary_search(sorted_.begin(), sorted_.end(), id);
  return negation_ ? !found : found;
}

bool SortFilterFunctor::Check(int64_t id) {
  int64_t lo = 0, hi = (int64_t)ids_.size() - 1;
  bool found = false;
  while (lo <= hi) {  // binary search, vector_index.h:117-133
    int64_t mid = (lo + hi) / 2;
    if (ids_[mid] == id) {
      found = true;
      break;
    }
    if (id < ids_[mid]) hi = mid - 1; else lo = mid + 1;
  }
  return negation_ ? !found : found;
}

namespace {

Status from_dg(dg_status st) {
  if (st == DG_OK) return Status::OK();
  char buf[512];
  dg_last_error(buf, sizeof(buf));
  int code;
  switch (st) {
    case DG_EINVAL: code = kEillegalParamteters; break;
    case DG_ENOT_TRAINED: code = kEVectorNotTrain; break;
    case DG_ENOT_SUPPORT: code = kEVectorNotSupport; break;
    case DG_EID_DUPLICATED: code = kEVectorIdDuplicated; break;
    case DG_ENOT_FOUND: code = kEVectorInvalid; break;
    default: code = kEInternal; break;
  }
  return {code, buf};
}

// pack VectorWithId batch -> contiguous floats + ids, checking dimensions
// (CheckVectorDimension semantics, utils.cc; normalization happens inside
// the library for cosine)
Status pack(const std::vector<VectorWithId>& v, int32_t dim,
            std::vector<float>& x, std::vector<int64_t>& ids) {
  x.resize((size_t)v.size() * dim);
  ids.resize(v.size());
  for (size_t i = 0; i < v.size(); i++) {
    if ((int32_t)v[i].vector.float_values.size() != dim)
      return {kEillegalParamteters, "dimension not match"};
    memcpy(&x[i * dim], v[i].vector.float_values.data(),
           (size_t)dim * sizeof(float));
    ids[i] = v[i].id;
  }
  return Status::OK();
}

class GpuIndexBase : public VectorIndex {
 public:
  GpuIndexBase(dg_index_kind kind, MetricType metric, int32_t dim,
               int32_t nlist, int device, int32_t pq_m = 0) {
    metric_ = metric;
    dim_ = dim;
    dg_index_desc desc{};
    desc.kind = (int32_t)kind;
    desc.metric = (int32_t)metric;
    desc.d = dim;
    desc.nlist = nlist;
    desc.pq_m = pq_m;
    desc.pq_nbits = 8;
    desc.device = device;
    create_st_ = dg_index_create(&idx_, &desc);
  }
  ~GpuIndexBase() override {
    if (idx_) dg_index_destroy(idx_);
  }
  Status CreateStatus() const { return from_dg(create_st_); }

  int32_t GetDimension() override { return dim_; }
  MetricType GetMetricType() override { return metric_; }
  Status GetCount(int64_t& count) override {
    dg_stats_out s{};
    dg_status st = dg_stats(idx_, &s);
    count = s.ntotal;
    return from_dg(st);
  }
  Status GetMemorySize(int64_t& bytes) override {
    dg_stats_out s{};
    dg_status st = dg_stats(idx_, &s);
    bytes = s.device_bytes;
    return from_dg(st);
  }
  Status Add(const std::vector<VectorWithId>& v) override {
    if (v.empty()) return {kEillegalParamteters, "vector_with_ids is empty"};
    std::vector<float> x;
    std::vector<int64_t> ids;
    Status s = pack(v, dim_, x, ids);
    if (!s.ok()) return s;
    return from_dg(dg_add(idx_, (int64_t)v.size(), ids.data(), x.data()));
  }
  Status Upsert(const std::vector<VectorWithId>& v) override {
    if (v.empty()) return {kEillegalParamteters, "vector_with_ids is empty"};
    std::vector<float> x;
    std::vector<int64_t> ids;
    Status s = pack(v, dim_, x, ids);
    if (!s.ok()) return s;
    return from_dg(dg_upsert(idx_, (int64_t)v.size(), ids.data(), x.data()));
  }
  Status Delete(const std::vector<int64_t>& ids) override {
    if (ids.empty()) return {kEillegalParamteters, "empty ids"};
    return from_dg(dg_remove(idx_, (int64_t)ids.size(), ids.data()));
  }
  Status Train(const std::vector<VectorWithId>& v) override {
    if (v.empty()) return {kEillegalParamteters, "data size invalid"};
    std::vector<float> x;
    std::vector<int64_t> ids;
    Status s = pack(v, dim_, x, ids);
    if (!s.ok()) return s;
    s = from_dg(dg_train(idx_, (int64_t)v.size(), x.data()));
    if (s.ok()) train_data_size_ = (int64_t)v.size();
    return s;
  }
  Status Train(std::vector<float>& train_datas) override {
    // raw-float Train (vector_index.h:196; size checks per
    // vector_index_ivf_flat.cc:644-652)
    size_t data_size = train_datas.size() / dim_;
    if (data_size == 0)
      return {kEillegalParamteters, "data size invalid"};
    if (train_datas.size() % dim_ != 0)
      return {kEillegalParamteters, "dimension not match"};
    Status s = from_dg(dg_train(idx_, (int64_t)data_size,
                                train_datas.data()));
    if (s.ok()) train_data_size_ = (int64_t)data_size;
    return s;
  }
  bool IsTrained() override {
    dg_stats_out s{};
    return dg_stats(idx_, &s) == DG_OK && s.is_trained;
  }
  Status GetDeletedCount(int64_t& deleted_count) override {
    // the reference's faiss path compacts on remove and reports 0
    // (vector_index_ivf_flat.cc:539-542); here removes tombstone until the
    // next finalize, so the pending count is reported
    dg_stats_out s{};
    dg_status st = dg_stats(idx_, &s);
    deleted_count = s.deleted_count;
    return from_dg(st);
  }
  bool IsExceedsMaxElements(int64_t) override {
    return false;  // flat.cc:510-512, ivf_flat.cc:573-575
  }
  bool NeedToRebuild() override { return false; }  // per-kind overrides
  bool NeedToSave(int64_t last_save_log_behind) override {
    // flat.cc:515-531 / ivf_flat.cc:786-800: trained, non-empty, and the
    // raft log has run ahead of the last snapshot by > need_save_count
    // (DEFINE_int64(..._need_save_count, 10000) in all three indexes)
    dg_stats_out s{};
    if (dg_stats(idx_, &s) != DG_OK) return false;
    if (!s.is_trained || s.ntotal == 0) return false;
    return last_save_log_behind > kNeedSaveCount;
  }
  bool SupportSave() override { return true; }
  void LockWrite() override { dg_lock_write(idx_); }
  void UnlockWrite() override { dg_unlock_write(idx_); }
  Status Save(const std::string& path) override {
    // the snapshot cycle ships faiss::write_index files
    // (vector_index_snapshot_manager.cc:583-599) — emit the compatible
    // container so CPU nodes can ingest the snapshot
    return from_dg(dg_save_faiss(idx_, path.c_str()));
  }
  Status Load(const std::string& path) override {
    dg_index* ni = nullptr;
    dg_status st = dg_load_faiss(&ni, path.c_str(), (int32_t)metric_, -1);
    if (st != DG_OK) return from_dg(st);
    dg_index_destroy(idx_);
    idx_ = ni;
    return Status::OK();
  }

  Status Search(const std::vector<VectorWithId>& queries, uint32_t topk,
                const std::vector<std::shared_ptr<FilterFunctor>>& filters,
                bool, const VectorSearchParameter& p,
                std::vector<VectorWithDistanceResult>& results) override {
    // mirrors VectorIndexFlat/IvfFlat::Search argument handling
    // (vector_index_flat.cc:205-221, ivf_flat.cc:191-236)
    if (queries.empty())
      return {kEillegalParamteters, "vector_with_ids is empty"};
    if (topk == 0) return Status::OK();
    std::vector<float> x;
    std::vector<int64_t> ids;
    Status s = pack(queries, dim_, x, ids);
    if (!s.ok()) return s;

    dg_filter df{};
    dg_filter* dfp = nullptr;
    if (!filters.empty()) {
      // one translatable filter supported natively; otherwise the reader's
      // post-filter path applies (not replicated here)
      if (filters.size() == 1 && filters[0]->ToDeviceFilter(&df)) {
        dfp = &df;
      } else {
        return {kEVectorNotSupport, "composite filters via reader fallback"};
      }
    }
    int32_t nprobe = p.ivf_flat_nprobe;
    std::vector<float> dist((size_t)queries.size() * topk);
    std::vector<int64_t> labels((size_t)queries.size() * topk, -1);
    dg_status st = dg_search(idx_, (int64_t)queries.size(), x.data(),
                             (int32_t)topk, nprobe, dfp, dist.data(),
                             labels.data());
    if (st != DG_OK) return from_dg(st);
    // FillSearchResult shaping (utils.cc:612-655)
    results.clear();
    results.resize(queries.size());
    for (size_t row = 0; row < queries.size(); row++) {
      for (uint32_t i = 0; i < topk; i++) {
        size_t pos = row * topk + i;
        if (labels[pos] < 0) continue;  // padding skipped, :622-624
        VectorWithDistance vd;
        vd.vector_with_id.id = labels[pos];
        vd.vector_with_id.vector.dimension = dim_;
        vd.metric_type = metric_;
        vd.distance = (metric_ == MetricType::kL2)
                          ? dist[pos]
                          : 1.0f - dist[pos];  // the flip, :634
        results[row].vector_with_distances.push_back(std::move(vd));
      }
    }
    return Status::OK();
  }

  Status RangeSearch(const std::vector<VectorWithId>& queries, float radius,
                     const std::vector<std::shared_ptr<FilterFunctor>>& fs,
                     bool, const VectorSearchParameter&,
                     std::vector<VectorWithDistanceResult>& results) override {
    // mirrors VectorIndexIvfFlat::RangeSearch: for IP/cosine the dingo
    // radius converts to a faiss score threshold 1 - r
    // (src/vector/vector_index_ivf_flat.cc:302-305)
    if (queries.empty())
      return {kEillegalParamteters, "vector_with_ids is empty"};
    std::vector<float> x;
    std::vector<int64_t> ids;
    Status s = pack(queries, dim_, x, ids);
    if (!s.ok()) return s;
    dg_filter df{};
    dg_filter* dfp = nullptr;
    if (!fs.empty()) {
      if (fs.size() == 1 && fs[0]->ToDeviceFilter(&df)) dfp = &df;
      else return {kEVectorNotSupport, "composite filters via reader"};
    }
    float r = metric_ == MetricType::kL2 ? radius : 1.0f - radius;
    std::vector<int64_t> lims(queries.size() + 1);
    int64_t* rids = nullptr;
    float* rdists = nullptr;
    dg_status st = dg_range_search(idx_, (int64_t)queries.size(), x.data(),
                                   r, dfp, lims.data(), &rids, &rdists);
    if (st != DG_OK) return from_dg(st);
    results.clear();
    results.resize(queries.size());
    for (size_t q = 0; q < queries.size(); q++) {
      for (int64_t i = lims[q]; i < lims[q + 1]; i++) {
        VectorWithDistance vd;
        vd.vector_with_id.id = rids[i];
        vd.vector_with_id.vector.dimension = dim_;
        vd.metric_type = metric_;
        vd.distance = (metric_ == MetricType::kL2) ? rdists[i]
                                                   : 1.0f - rdists[i];
        results[q].vector_with_distances.push_back(std::move(vd));
      }
    }
    dg_free(rids);
    dg_free(rdists);
    return Status::OK();
  }

 protected:
  static constexpr int64_t kNeedSaveCount = 10000;  // *_need_save_count
  static constexpr int64_t kMaxPointsPerCentroid = 256;  // faiss Clustering
  dg_index* idx_ = nullptr;
  dg_status create_st_ = DG_OK;
  MetricType metric_;
  int32_t dim_;
  int64_t train_data_size_ = 0;
};

class GpuFlatIndex : public GpuIndexBase {
 public:
  GpuFlatIndex(MetricType m, int32_t d, int dev)
      : GpuIndexBase(DG_INDEX_FLAT, m, d, 0, dev) {}
  bool NeedTrain() override { return false; }
  // Flat has no NeedToRebuild trigger in the reference (base default false)
};

class GpuIvfFlatIndex : public GpuIndexBase {
 public:
  GpuIvfFlatIndex(MetricType m, int32_t d, int32_t nlist, int dev)
      : GpuIndexBase(DG_INDEX_IVF_FLAT, m, d, nlist, dev),
        nlist_org_(nlist) {}
  bool NeedTrain() override { return !IsTrained(); }
  bool NeedToRebuild() override {
    // restates vector_index_ivf_flat.cc:751-783: rebuild when the data has
    // outgrown the structure (degraded nlist, or the train sample is now
    // under half the data) past max_points_per_centroid * nlist
    dg_stats_out s{};
    if (dg_stats(idx_, &s) != DG_OK || !s.is_trained) return false;
    const int64_t nlist_now = s.nlist;  // degrades to 1 at small trains
    if (nlist_now == nlist_org_ && nlist_now == 1) return false;
    const bool grown = s.ntotal >= kMaxPointsPerCentroid * nlist_org_;
    if (nlist_now != nlist_org_ && nlist_now == 1 && grown) return true;
    if (nlist_now == nlist_org_ && nlist_now != 1 && grown)
      return train_data_size_ <= s.ntotal / 2;
    return false;
  }

 private:
  int64_t nlist_org_;
};

class GpuIvfPqIndex : public GpuIndexBase {
 public:
  GpuIvfPqIndex(MetricType m, int32_t d, int32_t nlist, int32_t pq_m,
                int dev)
      : GpuIndexBase(DG_INDEX_IVF_PQ, m, d, nlist, dev, pq_m) {}
  bool NeedTrain() override { return !IsTrained(); }
  bool NeedToRebuild() override {
    // vector_index_raw_ivf_pq.cc:518-526
    dg_stats_out s{};
    if (dg_stats(idx_, &s) != DG_OK || !s.is_trained) return false;
    return (s.ntotal / 2) >= train_data_size_;
  }
};

}  // namespace


// ---- reader brute-force fallback (vector_reader.cc:1873-2048) ----
namespace {
struct HeapEntry {  // DistanceResult analog (vector_reader.cc:1845-1858)
  float distance;
  VectorWithDistance vd;
  bool operator<(const HeapEntry& o) const { return distance < o.distance; }
};
}  // namespace

Status BruteForceSearch(MetricType metric, int32_t dimension,
                        const RowIterator& next,
                        const std::vector<VectorWithId>& queries,
                        uint32_t topk,
                        const std::vector<std::shared_ptr<FilterFunctor>>& f,
                        const VectorSearchParameter& p,
                        std::vector<VectorWithDistanceResult>& results,
                        int64_t batch_count) {
  if (dimension <= 0) return {kEVectorInvalid, "dimension invalid"};
  std::vector<std::priority_queue<HeapEntry>> tops(queries.size());
  std::vector<VectorWithId> batch;
  batch.reserve(batch_count);
  auto flush = [&]() -> Status {
    if (batch.empty()) return Status::OK();
    // throwaway per-batch Flat index (vector_reader.cc:1939-1942)
    auto flat = NewFlatIndex(metric, dimension);
    if (!flat) return {kEInternal, "flat index create failed"};
    Status s = flat->Add(batch);
    if (!s.ok()) return s;
    std::vector<VectorWithDistanceResult> rb;
    s = flat->Search(queries, topk, f, false, p, rb);
    if (!s.ok()) return s;
    for (size_t i = 0; i < rb.size(); i++) {
      auto& top = tops[i];
      for (auto& vd : rb[i].vector_with_distances) {
        if (top.size() < topk) {
          top.push({vd.distance, vd});
        } else if (top.top().distance > vd.distance) {
          top.pop();
          top.push({vd.distance, vd});
        }
      }
    }
    batch.clear();
    return Status::OK();
  };
  VectorWithId row;
  while (next(&row)) {
    batch.push_back(std::move(row));
    if ((int64_t)batch.size() == batch_count) {
      Status s = flush();
      if (!s.ok()) return s;
    }
  }
  Status s = flush();
  if (!s.ok()) return s;
  // ascending by distance (deque emplace_front, vector_reader.cc:2032-2044)
  results.clear();
  results.resize(queries.size());
  for (size_t i = 0; i < tops.size(); i++) {
    auto& top = tops[i];
    std::vector<VectorWithDistance> tmp;
    while (!top.empty()) {
      tmp.push_back(top.top().vd);
      top.pop();
    }
    results[i].vector_with_distances.assign(tmp.rbegin(), tmp.rend());
  }
  return Status::OK();
}

Status SearchWithBruteForceFallback(
    VectorIndex* index, const std::function<RowIterator()>& scan_factory,
    const std::vector<VectorWithId>& queries, uint32_t topk,
    const std::vector<std::shared_ptr<FilterFunctor>>& f,
    const VectorSearchParameter& p,
    std::vector<VectorWithDistanceResult>& results) {
  Status s = index->Search(queries, topk, f, false, p, results);
  if (s.code == kEVectorNotSupport) {
    // the reader's fallback trigger (vector_reader.cc:1828-1831)
    return BruteForceSearch(index->GetMetricType(), index->GetDimension(),
                            scan_factory(), queries, topk, f, p, results);
  }
  return s;
}

std::unique_ptr<VectorIndex> NewFlatIndex(MetricType metric, int32_t dim,
                                          int device) {
  auto p = std::make_unique<GpuFlatIndex>(metric, dim, device);
  if (!p->CreateStatus().ok()) return nullptr;
  return p;
}

std::unique_ptr<VectorIndex> NewIvfFlatIndex(MetricType metric, int32_t dim,
                                             int32_t ncentroids, int device) {
  auto p = std::make_unique<GpuIvfFlatIndex>(metric, dim, ncentroids, device);
  if (!p->CreateStatus().ok()) return nullptr;
  return p;
}

std::unique_ptr<VectorIndex> NewIvfPqIndex(MetricType metric, int32_t dim,
                                           int32_t ncentroids,
                                           int32_t nsubvector, int device) {
  auto p = std::make_unique<GpuIvfPqIndex>(metric, dim, ncentroids,
                                           nsubvector, device);
  if (!p->CreateStatus().ok()) return nullptr;
  return p;
}

}  // namespace dingogpu

// ---------------- self-test (GPU) ----------------
extern "C" int dg_mirror_selftest(void) {
  using namespace dingogpu;
#define DG_ST(x) fprintf(stderr, "[selftest] %s\n", x)
  DG_ST("flat loop");
  const int32_t d = 64, n = 500;
  for (MetricType m :
       {MetricType::kL2, MetricType::kInnerProduct, MetricType::kCosine}) {
    auto idx = NewFlatIndex(m, d);
    if (!idx) return 1;
    std::vector<VectorWithId> batch(n);
    uint32_t s = 12345;
    for (int i = 0; i < n; i++) {
      batch[i].id = i * 7 + 3;
      batch[i].vector.dimension = d;
      batch[i].vector.float_values.resize(d);
      for (int j = 0; j < d; j++) {
        s = s * 1664525u + 1013904223u;
        batch[i].vector.float_values[j] = (s >> 8) * (1.0f / 16777216.0f);
      }
    }
    if (!idx->Add(batch).ok()) return 2;
    std::vector<VectorWithDistanceResult> res;
    VectorSearchParameter p;
    if (!idx->Search(batch, 3, {}, false, p, res).ok()) return 3;
    if (res.size() != (size_t)n) return 4;
    for (int i = 0; i < n; i++) {
      if (res[i].vector_with_distances.empty()) return 5;
      // self-top-1 (test_vector_index_recall_flat.cc:170-236) holds for L2
      // and cosine; NOT for raw inner product (the max-IP neighbor of x
      // need not be x itself)
      if (m != MetricType::kInnerProduct &&
          res[i].vector_with_distances[0].vector_with_id.id != batch[i].id)
        return 6;
      float dist = res[i].vector_with_distances[0].distance;
      // L2 self-dist 0; cosine reported as 1 - score -> 0 for self
      if (m == MetricType::kL2 && std::fabs(dist) > 1e-3f) return 7;
      if (m == MetricType::kCosine && std::fabs(dist) > 1e-3f) return 8;
    }
    // range filter: restrict to ids < 100
    std::vector<std::shared_ptr<FilterFunctor>> f{
        std::make_shared<RangeFilterFunctor>(0, 100)};
    std::vector<VectorWithDistanceResult> res2;
    if (!idx->Search({batch[0]}, 5, f, false, p, res2).ok()) return 9;
    for (auto& vd : res2[0].vector_with_distances)
      if (vd.vector_with_id.id >= 100) return 10;
  }

  DG_ST("ivf lifecycle");
  // ---- IVF lifecycle through the mirror (Train/Add/Search/RangeSearch/
  // Upsert/Delete/Save/Load), L2 ----
  {
    const int32_t d2 = 32, n2 = 2000;
    auto ivf = NewIvfFlatIndex(MetricType::kL2, d2, 16);
    if (!ivf) return 20;
    std::vector<VectorWithId> batch(n2);
    uint32_t s = 99;
    for (int i = 0; i < n2; i++) {
      batch[i].id = i;
      batch[i].vector.dimension = d2;
      batch[i].vector.float_values.resize(d2);
      for (int j = 0; j < d2; j++) {
        s = s * 1664525u + 1013904223u;
        batch[i].vector.float_values[j] = (s >> 8) * (1.0f / 16777216.0f);
      }
    }
    if (ivf->IsTrained()) return 21;           // NeedTrain before Train
    if (!ivf->NeedTrain()) return 22;
    // untrained Search: blank results, OK (ivf_flat.cc:223-227)
    std::vector<VectorWithDistanceResult> r0;
    VectorSearchParameter p;
    p.ivf_flat_nprobe = 16;
    if (!ivf->Search({batch[0]}, 3, {}, false, p, r0).ok()) return 23;
    if (!r0[0].vector_with_distances.empty()) return 24;
    if (!ivf->Train(batch).ok()) return 25;
    if (!ivf->IsTrained()) return 26;
    if (!ivf->Add(batch).ok()) return 27;
    int64_t cnt = 0;
    ivf->GetCount(cnt);
    if (cnt != n2) return 28;
    std::vector<VectorWithDistanceResult> r1;
    if (!ivf->Search({batch[7]}, 3, {}, false, p, r1).ok()) return 29;
    if (r1[0].vector_with_distances.empty() ||
        r1[0].vector_with_distances[0].vector_with_id.id != 7)
      return 30;
    DG_ST("range search");
    // range search around the self-distance
    std::vector<VectorWithDistanceResult> r2;
    if (!ivf->RangeSearch({batch[7]}, 0.5f, {}, false, p, r2).ok())
      return 31;
    bool has_self = false;
    for (auto& vd : r2[0].vector_with_distances)
      if (vd.vector_with_id.id == 7) has_self = true;
    if (!has_self) return 32;
    DG_ST("delete+upsert");
    // delete + upsert
    if (!ivf->Delete({7}).ok()) return 33;
    int64_t delc = -1;
    if (!ivf->GetDeletedCount(delc).ok() || delc != 1) return 61;
    std::vector<VectorWithDistanceResult> r3;
    ivf->Search({batch[7]}, 1, {}, false, p, r3);
    if (!r3[0].vector_with_distances.empty() &&
        r3[0].vector_with_distances[0].vector_with_id.id == 7)
      return 34;
    if (!ivf->Upsert({batch[7]}).ok()) return 35;
    std::vector<VectorWithDistanceResult> r4;
    ivf->Search({batch[7]}, 1, {}, false, p, r4);
    if (r4[0].vector_with_distances.empty() ||
        r4[0].vector_with_distances[0].vector_with_id.id != 7)
      return 36;
    DG_ST("save/load");
    // save / load round trip
    const char* path = "/tmp/dg_mirror_selftest.dgi";
    if (!ivf->Save(path).ok()) return 37;
    if (!ivf->Load(path).ok()) return 38;
    std::vector<VectorWithDistanceResult> r5;
    ivf->Search({batch[7]}, 1, {}, false, p, r5);
    if (r5[0].vector_with_distances.empty() ||
        r5[0].vector_with_distances[0].vector_with_id.id != 7)
      return 39;
    DG_ST("dim mismatch");
    // dimension mismatch rejected (CheckVectorDimension semantics)
    VectorWithId bad;
    bad.id = 999999;
    bad.vector.dimension = d2 / 2;
    bad.vector.float_values.resize(d2 / 2);
    if (ivf->Add({bad}).ok()) return 40;

    DG_ST("lifecycle virtuals");
    // ---- wrapper-lifecycle virtuals (§8b must-implement set) ----
    if (!ivf->SupportSave()) return 41;
    if (ivf->IsExceedsMaxElements(1 << 20)) return 42;  // always false
    int64_t delc2 = -1;  // faiss-file load dropped the tombstones
    if (!ivf->GetDeletedCount(delc2).ok() || delc2 != 0) return 43;
    if (ivf->NeedToSave(100)) return 44;      // behind <= need_save_count
    if (!ivf->NeedToSave(20000)) return 45;   // behind > 10000
    if (ivf->NeedToRebuild()) return 46;      // ntotal < 256 * nlist
    ivf->LockWrite();   // exclusive fork-save window lock
    ivf->UnlockWrite();
    DG_ST("raw-float train");
    // raw-float Train(std::vector<float>&) on a fresh index
    {
      auto ivf2 = NewIvfFlatIndex(MetricType::kL2, d2, 4);
      if (!ivf2) return 47;
      std::vector<float> tdata;
      tdata.reserve((size_t)n2 * d2);
      for (auto& vw : batch)
        tdata.insert(tdata.end(), vw.vector.float_values.begin(),
                     vw.vector.float_values.end());
      std::vector<float> badsize(tdata.begin(), tdata.begin() + d2 / 2);
      if (ivf2->Train(badsize).ok()) return 48;  // size % dim != 0 rejected
      if (!ivf2->Train(tdata).ok()) return 49;
      if (!ivf2->IsTrained()) return 50;
    }
    DG_ST("concrete filter");
    // ConcreteFilterFunctor (IDSelectorBatch semantics) incl. Build hook
    {
      std::vector<int64_t> want{3, 9, 15};
      auto cf = std::make_shared<ConcreteFilterFunctor>(want);
      std::vector<int64_t> idmap;
      cf->Build(idmap);  // reference default no-op hook, callable
      if (!cf->Check(9) || cf->Check(10)) return 51;
      std::vector<std::shared_ptr<FilterFunctor>> fs{cf};
      std::vector<VectorWithDistanceResult> r6;
      if (!ivf->Search({batch[3]}, 3, fs, false, p, r6).ok()) return 52;
      if (r6[0].vector_with_distances.empty()) return 53;
      for (auto& vd : r6[0].vector_with_distances) {
        int64_t id = vd.vector_with_id.id;
        if (id != 3 && id != 9 && id != 15) return 54;
      }
      // negated form excludes them
      auto nf = std::make_shared<ConcreteFilterFunctor>(want, true);
      std::vector<std::shared_ptr<FilterFunctor>> fs2{nf};
      std::vector<VectorWithDistanceResult> r7;
      if (!ivf->Search({batch[3]}, 5, fs2, false, p, r7).ok()) return 55;
      for (auto& vd : r7[0].vector_with_distances) {
        int64_t id = vd.vector_with_id.id;
        if (id == 3 || id == 9 || id == 15) return 56;
      }
    }
  }

  // ---- reader brute-force path + EVECTOR_NOT_SUPPORT fallback round trip
  // (vector_reader.cc:1873-2048, :1828-1831) ----
  DG_ST("brute force");
  {
    const int32_t d3 = 24;
    const int64_t n3 = 5000;  // > 2 batches of 2048
    std::vector<VectorWithId> rows(n3);
    uint32_t s = 777;
    for (int64_t i = 0; i < n3; i++) {
      rows[i].id = i * 2 + 1;
      rows[i].vector.dimension = d3;
      rows[i].vector.float_values.resize(d3);
      for (int j = 0; j < d3; j++) {
        s = s * 1664525u + 1013904223u;
        rows[i].vector.float_values[j] = (s >> 8) * (1.0f / 16777216.0f);
      }
    }
    std::vector<VectorWithId> queries(rows.begin(), rows.begin() + 4);
    auto make_scan = [&]() -> RowIterator {
      auto pos = std::make_shared<int64_t>(0);
      return [&rows, pos](VectorWithId* out) {
        if (*pos >= (int64_t)rows.size()) return false;
        *out = rows[(*pos)++];
        return true;
      };
    };
    VectorSearchParameter p3;
    // ground truth: one persistent Flat index over all rows
    auto all = NewFlatIndex(MetricType::kL2, d3);
    if (!all) return 70;
    if (!all->Add(rows).ok()) return 71;
    std::vector<VectorWithDistanceResult> want;
    if (!all->Search(queries, 7, {}, false, p3, want).ok()) return 72;
    // KV-scan brute force (2048-row batches + heap merge) must agree
    std::vector<VectorWithDistanceResult> got;
    if (!BruteForceSearch(MetricType::kL2, d3, make_scan(), queries, 7, {},
                          p3, got).ok())
      return 73;
    if (got.size() != want.size()) return 74;
    for (size_t i = 0; i < got.size(); i++) {
      auto& g = got[i].vector_with_distances;
      auto& w = want[i].vector_with_distances;
      if (g.size() != w.size()) return 75;
      for (size_t j = 0; j < g.size(); j++) {
        if (g[j].vector_with_id.id != w[j].vector_with_id.id) return 76;
        if (std::fabs(g[j].distance - w[j].distance) > 1e-4f) return 77;
      }
    }
    // NOT_SUPPORT fallback round trip: an index that rejects the request
    // must transparently drop to the brute-force scan
    class NotSupportIndex : public GpuFlatIndex {
     public:
      NotSupportIndex(MetricType m, int32_t d) : GpuFlatIndex(m, d, -1) {}
      Status Search(const std::vector<VectorWithId>&, uint32_t,
                    const std::vector<std::shared_ptr<FilterFunctor>>&,
                    bool, const VectorSearchParameter&,
                    std::vector<VectorWithDistanceResult>&) override {
        return {kEVectorNotSupport, "not support"};
      }
    };
    NotSupportIndex ns(MetricType::kL2, d3);
    std::vector<VectorWithDistanceResult> got2;
    if (!SearchWithBruteForceFallback(&ns, make_scan, queries, 7, {}, p3,
                                      got2).ok())
      return 78;
    if (got2.size() != want.size()) return 79;
    for (size_t i = 0; i < got2.size(); i++)
      if (got2[i].vector_with_distances.size() !=
              want[i].vector_with_distances.size() ||
          got2[i].vector_with_distances[0].vector_with_id.id !=
              want[i].vector_with_distances[0].vector_with_id.id)
        return 80;
  }
  DG_ST("done");
  return 0;
}
