// vector_index_gpu.h — C++ mirror of dingo-store's VectorIndex plugin
// surface (reference: src/vector/vector_index.h:56-279) implemented on top
// of the C-ABI in include/dingo_gpu.h.  A dingo-store maintainer drops this
// subclass into the Index role by registering it in VectorIndexFactory::New
// (src/vector/vector_index_factory.cc:40-95) — see INTEGRATION.md for the
// exact stub.  Proto schemas are absent from the reference tree
// (dingo-store-proto is an empty submodule), so the boundary types below are
// plain-struct re-declarations reconstructed from the call sites cited in
// SURVEY.md §3.1/§8b.
#pragma once

#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "../../include/dingo_gpu.h"

namespace dingogpu {

// ---- pb::common::* re-declarations (schema absent; field set from call
// sites, e.g. ExtractVectorValue src/vector/vector_index_utils.cc:564-609,
// FillSearchResult :612-655) ----
enum class MetricType { kL2 = 0, kInnerProduct = 1, kCosine = 2 };
enum class ValueType { kFloat = 0, kUint8 = 1 };

struct Vector {
  int32_t dimension = 0;
  ValueType value_type = ValueType::kFloat;
  std::vector<float> float_values;
};

struct VectorWithId {
  int64_t id = -1;
  Vector vector;
};

struct VectorWithDistance {
  VectorWithId vector_with_id;
  float distance = 0.f;  // dingo convention: L2 raw sqr; IP/cos 1 - score
  MetricType metric_type = MetricType::kL2;
};

struct VectorWithDistanceResult {
  std::vector<VectorWithDistance> vector_with_distances;
};

struct VectorSearchParameter {
  int32_t top_n = 0;
  bool use_brute_force = false;
  bool without_vector_data = true;
  int32_t ivf_flat_nprobe = 0;  // parameter.ivf_flat().nprobe()
  int32_t ivf_pq_nprobe = 0;
  bool enable_range_search = false;
  float radius = 0.f;
};

// butil::Status analog (pb::error::Errno codes preserved numerically where
// the reader dispatches on them)
struct Status {
  int code = 0;  // 0 = OK
  std::string msg;
  bool ok() const { return code == 0; }
  static Status OK() { return {}; }
};
// Errno values the reader dispatches on (reference pb/error.proto usage):
constexpr int kEillegalParamteters = 30001;  // EILLEGAL_PARAMTETERS
constexpr int kEVectorNotTrain = 30008;      // triggers train-first handling
constexpr int kEVectorNotSupport = 30010;    // reader brute-force fallback
constexpr int kEVectorIdDuplicated = 30011;
constexpr int kEVectorInvalid = 30012;
constexpr int kEInternal = 10002;

// ---- FilterFunctor mirror (vector_index.h:67-146) ----
class FilterFunctor {
 public:
  virtual ~FilterFunctor() = default;
  // hook the reference gives filters to pre-index the id universe
  // (vector_index.h:70); default no-op like the reference
  virtual void Build(std::vector<int64_t>& /*id_map*/) {}
  virtual bool Check(int64_t vector_id) = 0;
  // translate to the device filter; default: unsupported (caller falls back
  // to post-filtering like the reader's over-fetch path)
  virtual bool ToDeviceFilter(dg_filter* out) const { return false; }
};

class RangeFilterFunctor : public FilterFunctor {  // vector_index.h:77-84
 public:
  RangeFilterFunctor(int64_t min_id, int64_t max_id)
      : min_(min_id), max_(max_id) {}
  bool Check(int64_t id) override { return id >= min_ && id < max_; }
  bool ToDeviceFilter(dg_filter* out) const override {
    *out = {};
    out->kind = DG_FILTER_RANGE;
    out->min_id = min_;
    out->max_id = max_;
    return true;
  }

 private:
  int64_t min_, max_;
};

class SortFilterFunctor : public FilterFunctor {  // vector_index.h:109-146
 public:
  explicit SortFilterFunctor(std::vector<int64_t> ids, bool negation = false)
      : ids_(std::move(ids)), negation_(negation) {}
  bool Check(int64_t id) override;
  bool ToDeviceFilter(dg_filter* out) const override {
    *out = {};
    out->kind = DG_FILTER_SORTED_IDS;
    out->ids = ids_.data();
    out->n_ids = (int64_t)ids_.size();
    out->negate = negation_ ? 1 : 0;
    return true;
  }

 private:
  std::vector<int64_t> ids_;
  bool negation_;
};

// ConcreteFilterFunctor mirror (vector_index.h:86-106: faiss IDSelectorBatch
// membership + optional negation); device form = sorted-id filter
class ConcreteFilterFunctor : public FilterFunctor {
 public:
  explicit ConcreteFilterFunctor(const std::vector<int64_t>& ids,
                                 bool is_negation = false);
  bool Check(int64_t id) override;
  bool ToDeviceFilter(dg_filter* out) const override {
    *out = {};
    out->kind = DG_FILTER_SORTED_IDS;
    out->ids = sorted_.data();
    out->n_ids = (int64_t)sorted_.size();
    out->negate = negation_ ? 1 : 0;
    return true;
  }

 private:
  std::vector<int64_t> sorted_;
  bool negation_;
};

// ---- VectorIndex mirror (vector_index.h:148-229 virtuals; every
// must-implement virtual the wrapper/manager calls is present) ----
class VectorIndex {
 public:
  virtual ~VectorIndex() = default;
  virtual int32_t GetDimension() = 0;
  virtual MetricType GetMetricType() = 0;
  virtual Status GetCount(int64_t& count) = 0;
  virtual Status GetDeletedCount(int64_t& deleted_count) = 0;
  virtual Status GetMemorySize(int64_t& bytes) = 0;
  virtual bool IsExceedsMaxElements(int64_t vector_size) = 0;
  virtual Status Add(const std::vector<VectorWithId>& v) = 0;
  virtual Status Upsert(const std::vector<VectorWithId>& v) = 0;
  virtual Status Delete(const std::vector<int64_t>& ids) = 0;
  virtual Status Train(const std::vector<VectorWithId>& v) = 0;
  // raw float form (vector_index.h:196: Train(std::vector<float>&))
  virtual Status Train(std::vector<float>& train_datas) = 0;
  virtual bool IsTrained() = 0;
  virtual bool NeedTrain() = 0;
  // rebuild/save scheduling hooks VectorIndexManager drives
  // (vector_index.h:198-204; semantics per concrete index)
  virtual bool NeedToRebuild() = 0;
  virtual bool NeedToSave(int64_t last_save_log_behind) = 0;
  virtual bool SupportSave() { return false; }
  // exclusive lock for the wrapper's fork-save window
  // (vector_index.h:192-193; rw_lock_.LockWrite/UnlockWrite)
  virtual void LockWrite() = 0;
  virtual void UnlockWrite() = 0;
  virtual Status Save(const std::string& path) = 0;
  virtual Status Load(const std::string& path) = 0;
  virtual Status Search(const std::vector<VectorWithId>& queries,
                        uint32_t topk,
                        const std::vector<std::shared_ptr<FilterFunctor>>& f,
                        bool reconstruct, const VectorSearchParameter& p,
                        std::vector<VectorWithDistanceResult>& results) = 0;
  virtual Status RangeSearch(
      const std::vector<VectorWithId>& queries, float radius,
      const std::vector<std::shared_ptr<FilterFunctor>>& f, bool reconstruct,
      const VectorSearchParameter& p,
      std::vector<VectorWithDistanceResult>& results) = 0;
};

// ---- reader brute-force fallback ----
// Mirror of VectorReader::BruteForceSearch (vector_reader.cc:1873-2048):
// scan the region's rows in FLAGS_vector_index_bruteforce_batch_count
// batches (default 2048, vector_reader.cc:61), build a throwaway Flat
// index per batch, search it, and merge per-query top-k by the dingo
// distance (max-heap keeps the smallest topk; output ascending).  The KV
// iterator is abstracted as a pull function (the real reader wires the
// RocksDB range scan + proto decode here, vector_reader.cc:1922-1936).
using RowIterator = std::function<bool(VectorWithId*)>;

Status BruteForceSearch(MetricType metric, int32_t dimension,
                        const RowIterator& next,
                        const std::vector<VectorWithId>& queries,
                        uint32_t topk,
                        const std::vector<std::shared_ptr<FilterFunctor>>& f,
                        const VectorSearchParameter& p,
                        std::vector<VectorWithDistanceResult>& results,
                        int64_t batch_count = 2048);

// Search with the reader's EVECTOR_NOT_SUPPORT fallback round trip
// (vector_reader.cc:1828-1831): an index that rejects the request drops
// to the KV-scan brute force.  scan_factory yields a fresh iterator.
Status SearchWithBruteForceFallback(
    VectorIndex* index, const std::function<RowIterator()>& scan_factory,
    const std::vector<VectorWithId>& queries, uint32_t topk,
    const std::vector<std::shared_ptr<FilterFunctor>>& f,
    const VectorSearchParameter& p,
    std::vector<VectorWithDistanceResult>& results);

// GPU-backed concrete indexes (Flat / IVF-Flat), the factory, and a
// self-test used by the GPU test suite.
std::unique_ptr<VectorIndex> NewFlatIndex(MetricType metric, int32_t dim,
                                          int device = -1);
std::unique_ptr<VectorIndex> NewIvfFlatIndex(MetricType metric, int32_t dim,
                                             int32_t ncentroids,
                                             int device = -1);
std::unique_ptr<VectorIndex> NewIvfPqIndex(MetricType metric, int32_t dim,
                                           int32_t ncentroids,
                                           int32_t nsubvector,
                                           int device = -1);

}  // namespace dingogpu

extern "C" {
// Runs a tiny Flat + IVF search through the C++ plugin mirror on the GPU and
// verifies the dingo-store observable semantics (1 - score flip for
// IP/cosine, vector_index_utils.cc:634; self-top-1).  Returns 0 on success.
int dg_mirror_selftest(void);
}
