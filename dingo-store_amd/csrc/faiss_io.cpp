// faiss_io.cpp — faiss-1.7.x-compatible index containers (SURVEY.md §8f
// rank 1).  The reference's snapshot/install cycle ships faiss::write_index
// files (src/vector/vector_index_snapshot_manager.cc:583-599); a drop-in
// Index role must interchange snapshots with CPU nodes, so this file
// implements the published faiss container byte layout for exactly the
// three index shapes the reference constructs:
//   FLAT     IndexIDMap2{IndexFlatL2|IndexFlatIP}  (vector_index_flat.cc:81-98)
//   IVF_FLAT IndexIVFFlat                          (vector_index_ivf_flat.cc:805-823)
//   IVF_PQ   IndexIVFPQ (by_residual, nbits=8)     (vector_index_raw_ivf_pq.cc:551-570)
//
// Layout (faiss 1.7.x index_write.cpp / index_read.cpp, restated from the
// published format — the reference's faiss fork is >= 1.7.3 since it uses
// faiss::SearchParameters at vector_index_flat.cc:232):
//   write_index_header: d:i32, ntotal:i64, dummy:i64=1<<20 (x2),
//                       is_trained:u8, metric:i32 (+metric_arg:f32 iff >1)
//   vector<T>:          count:u64 (elements), raw data
//   IndexFlat payload:  "xb vector" — count:u64 in FLOATS (codes bytes / 4),
//                       then ntotal*d fp32 (legacy-compatible encoding)
//   IndexIDMap2:        "IxM2" hdr, inner index, id_map as vector<i64>
//   IndexIVF header:    index header, nlist:u64, nprobe:u64, quantizer
//                       (IndexFlat holding centroids), direct_map
//                       (type:u8=0 NoMap + empty vector)
//   InvertedLists:      "ilar", nlist:u64, code_size:u64, "full",
//                       sizes vector<u64>, then per nonempty list:
//                       codes (n*code_size bytes) then ids (n x i64)
//   IndexIVFPQ:         "IwPQ", ivf header, by_residual:u8(bool),
//                       code_size:u64, PQ {d:u64, M:u64, nbits:u64,
//                       centroids vector<f32>}, inverted lists
// All integers little-endian, no alignment padding (faiss fwrite layout).
// Cosine indexes are written as IP over stored normalized vectors, exactly
// like the reference (vector_index_flat.cc:88-91).
#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <vector>

#include "dg_internal.h"

namespace {

constexpr uint32_t fcc(const char s[5]) {
  return (uint32_t)(uint8_t)s[0] | ((uint32_t)(uint8_t)s[1] << 8) |
         ((uint32_t)(uint8_t)s[2] << 16) | ((uint32_t)(uint8_t)s[3] << 24);
}

struct FWriter {
  FILE* f = nullptr;
  bool ok = true;
  void raw(const void* p, size_t n) {
    if (ok && fwrite(p, 1, n, f) != n) ok = false;
  }
  void u8(uint8_t v) { raw(&v, 1); }
  void i32(int32_t v) { raw(&v, 4); }
  void u32(uint32_t v) { raw(&v, 4); }
  void i64(int64_t v) { raw(&v, 8); }
  void u64(uint64_t v) { raw(&v, 8); }
};

struct FReader {
  FILE* f = nullptr;
  bool ok = true;
  void raw(void* p, size_t n) {
    if (ok && fread(p, 1, n, f) != n) ok = false;
  }
  uint8_t u8() { uint8_t v = 0; raw(&v, 1); return v; }
  int32_t i32() { int32_t v = 0; raw(&v, 4); return v; }
  uint32_t u32() { uint32_t v = 0; raw(&v, 4); return v; }
  int64_t i64() { int64_t v = 0; raw(&v, 8); return v; }
  uint64_t u64() { uint64_t v = 0; raw(&v, 8); return v; }
};

// faiss MetricType: 0 = INNER_PRODUCT, 1 = L2
int to_faiss_metric(int32_t m) { return m == DG_METRIC_L2 ? 1 : 0; }

void write_index_header(FWriter& w, int32_t d, int64_t ntotal, bool trained,
                        int faiss_metric) {
  w.i32(d);
  w.i64(ntotal);
  w.i64(1 << 20);  // dummy (faiss legacy fields)
  w.i64(1 << 20);
  w.u8(trained ? 1 : 0);
  w.i32(faiss_metric);
  // metric_arg written only for metric > 1 (never for L2/IP)
}

struct IdxHeader {
  int32_t d;
  int64_t ntotal;
  bool trained;
  int32_t metric;  // faiss metric
};

bool read_index_header(FReader& r, IdxHeader& h) {
  h.d = r.i32();
  h.ntotal = r.i64();
  (void)r.i64();
  (void)r.i64();
  h.trained = r.u8() != 0;
  h.metric = r.i32();
  if (h.metric > 1) (void)r.u32();  // metric_arg (f32)
  return r.ok && h.d > 0 && h.ntotal >= 0;
}

// IndexFlat payload: xb-vector encoding (count in floats, data = raw fp32)
void write_flat_index(FWriter& w, int32_t d, int64_t n, int faiss_metric,
                      const float* data) {
  w.u32(faiss_metric == 1 ? fcc("IxF2") : fcc("IxFI"));
  write_index_header(w, d, n, true, faiss_metric);
  w.u64((uint64_t)n * d);  // count in floats
  w.raw(data, (size_t)n * d * 4);
}

bool read_flat_index(FReader& r, IdxHeader& h, std::vector<float>& data) {
  uint32_t h4 = r.u32();
  if (h4 != fcc("IxF2") && h4 != fcc("IxFI")) return false;
  if (!read_index_header(r, h)) return false;
  h.metric = (h4 == fcc("IxF2")) ? 1 : 0;
  uint64_t count = r.u64();
  if (!r.ok || count != (uint64_t)h.ntotal * h.d) return false;
  data.resize(count);
  r.raw(data.data(), count * 4);
  return r.ok;
}

void write_direct_map(FWriter& w) {
  w.u8(0);   // DirectMap::NoMap
  w.u64(0);  // empty array
}

bool read_direct_map(FReader& r) {
  uint8_t type = r.u8();
  uint64_t n = r.u64();
  if (type == 0 || type == 1) {
    // NoMap stores an empty array; Array stores ntotal entries
    std::vector<int64_t> skip(n);
    if (n) r.raw(skip.data(), n * 8);
    return r.ok;
  }
  if (type == 2) {  // Hashtable: vector of (key,id) pairs
    std::vector<int64_t> skip(n);
    if (n) r.raw(skip.data(), n * 8);
    return r.ok;
  }
  return false;
}

// host copies of the index data needed for writing
struct HostRows {
  std::vector<float> vectors;   // arrival order (FLAT/IVF)
  std::vector<uint8_t> codes;   // arrival order (PQ)
  std::vector<int64_t> ids;     // arrival order, -2 = tombstone
  std::vector<int32_t> assign;  // arrival order
};

dg_status download_rows(dg_index* ix, HostRows& h) {
  const int64_t n = ix->ntotal;
  const int32_t du = ix->d_user;   // file stride = caller's dimension
  const int32_t dp = ix->desc.d;   // device stride (padded)
  const bool is_pq = ix->desc.kind == DG_INDEX_IVF_PQ;
  h.ids.resize(n);
  h.assign.resize(n);
  if (n == 0) return DG_OK;
  if (hipMemcpy(h.ids.data(), ix->d_ids.p, (size_t)n * 8,
                hipMemcpyDeviceToHost) != hipSuccess)
    return DG_EINTERNAL;
  if (hipMemcpy(h.assign.data(), ix->d_assign.p, (size_t)n * 4,
                hipMemcpyDeviceToHost) != hipSuccess)
    return DG_EINTERNAL;
  if (is_pq) {
    h.codes.resize((size_t)n * ix->desc.pq_m);
    if (hipMemcpy(h.codes.data(), ix->d_codes.p, h.codes.size(),
                  hipMemcpyDeviceToHost) != hipSuccess)
      return DG_EINTERNAL;
  } else {
    h.vectors.resize((size_t)n * du);
    if (hipMemcpy2D(h.vectors.data(), (size_t)du * 4, ix->d_vectors.p,
                    (size_t)dp * 4, (size_t)du * 4, (size_t)n,
                    hipMemcpyDeviceToHost) != hipSuccess)
      return DG_EINTERNAL;
  }
  return DG_OK;
}

// group kept (non-tombstoned) rows by list: returns per-list row indexes
void group_rows(const HostRows& h, int32_t nlist,
                std::vector<std::vector<int64_t>>& lists, int64_t* kept) {
  lists.assign(nlist, {});
  int64_t k = 0;
  for (int64_t i = 0; i < (int64_t)h.ids.size(); i++) {
    if (h.ids[i] < 0) continue;  // tombstone
    int32_t l = h.assign[i];
    if (l >= 0 && l < nlist) {
      lists[l].push_back(i);
      k++;
    }
  }
  *kept = k;
}

void write_invlists(FWriter& w, const HostRows& h,
                    const std::vector<std::vector<int64_t>>& lists,
                    size_t code_size, int32_t d, bool is_pq, int32_t M) {
  w.u32(fcc("ilar"));
  w.u64(lists.size());
  w.u64(code_size);
  w.u32(fcc("full"));
  w.u64(lists.size());  // sizes vector count
  for (auto& l : lists) w.u64(l.size());
  std::vector<uint8_t> cbuf;
  std::vector<int64_t> ibuf;
  for (auto& l : lists) {
    if (l.empty()) continue;
    cbuf.resize(l.size() * code_size);
    ibuf.resize(l.size());
    for (size_t j = 0; j < l.size(); j++) {
      int64_t row = l[j];
      if (is_pq)
        memcpy(cbuf.data() + j * code_size, h.codes.data() + row * M,
               code_size);
      else
        memcpy(cbuf.data() + j * code_size,
               h.vectors.data() + (size_t)row * d, code_size);
      ibuf[j] = h.ids[row];
    }
    w.raw(cbuf.data(), cbuf.size());
    w.raw(ibuf.data(), ibuf.size() * 8);
  }
}

struct InvLists {
  uint64_t nlist = 0, code_size = 0;
  std::vector<std::vector<uint8_t>> codes;
  std::vector<std::vector<int64_t>> ids;
};

bool read_invlists(FReader& r, InvLists& il) {
  if (r.u32() != fcc("ilar")) return false;
  il.nlist = r.u64();
  il.code_size = r.u64();
  if (!r.ok || il.nlist > (1u << 24) || il.code_size > (1u << 20))
    return false;
  uint32_t lt = r.u32();
  std::vector<uint64_t> sizes(il.nlist, 0);
  if (lt == fcc("full")) {
    uint64_t cnt = r.u64();
    if (cnt != il.nlist) return false;
    r.raw(sizes.data(), il.nlist * 8);
  } else if (lt == fcc("sprs")) {
    uint64_t cnt = r.u64();  // count of u64s: pairs of (idx, size)
    std::vector<uint64_t> pairs(cnt);
    r.raw(pairs.data(), cnt * 8);
    for (uint64_t j = 0; j + 1 < cnt; j += 2) {
      if (pairs[j] >= il.nlist) return false;
      sizes[pairs[j]] = pairs[j + 1];
    }
  } else {
    return false;
  }
  if (!r.ok) return false;
  il.codes.resize(il.nlist);
  il.ids.resize(il.nlist);
  for (uint64_t l = 0; l < il.nlist; l++) {
    uint64_t n = sizes[l];
    if (!n) continue;
    il.codes[l].resize(n * il.code_size);
    il.ids[l].resize(n);
    r.raw(il.codes[l].data(), il.codes[l].size());
    r.raw(il.ids[l].data(), n * 8);
    if (!r.ok) return false;
  }
  return true;
}

struct DeviceGuard {
  int prev = -1;
  explicit DeviceGuard(int dev) {
    (void)hipGetDevice(&prev);
    if (dev >= 0 && dev != prev) (void)hipSetDevice(dev);
  }
  ~DeviceGuard() {
    if (prev >= 0) (void)hipSetDevice(prev);
  }
};

}  // namespace

extern "C" dg_status dg_save_faiss(dg_index* ix, const char* path) {
  if (!ix || !path) return DG_EINVAL;
  std::shared_lock lk(ix->rw);
  DeviceGuard g{ix->device};
  const int32_t d = ix->d_user;  // containers speak the caller's dimension
  const int fm = to_faiss_metric(ix->desc.metric);
  const bool is_pq = ix->desc.kind == DG_INDEX_IVF_PQ;
  const bool is_flat = ix->desc.kind == DG_INDEX_FLAT;
  if (!is_flat && !ix->trained) {
    dg_set_error("faiss save of untrained IVF index unsupported");
    return DG_ENOT_SUPPORT;
  }
  HostRows h;
  dg_status st = download_rows(ix, h);
  if (st != DG_OK) return st;

  FILE* f = fopen(path, "wb");
  if (!f) {
    dg_set_error("cannot open %s", path);
    return DG_EIO;
  }
  FWriter w{f};
  if (is_flat) {
    // IndexIDMap2 { IndexFlat }: inner codes = kept vectors in arrival
    // order; id_map = their ids (vector_index_flat.cc:81-98)
    std::vector<float> kept;
    std::vector<int64_t> idmap;
    kept.reserve(h.vectors.size());
    for (int64_t i = 0; i < ix->ntotal; i++) {
      if (h.ids[i] < 0) continue;
      kept.insert(kept.end(), h.vectors.begin() + (size_t)i * d,
                  h.vectors.begin() + (size_t)(i + 1) * d);
      idmap.push_back(h.ids[i]);
    }
    const int64_t n = (int64_t)idmap.size();
    w.u32(fcc("IxM2"));
    write_index_header(w, d, n, true, fm);
    write_flat_index(w, d, n, fm, kept.data());
    w.u64((uint64_t)n);  // id_map vector
    w.raw(idmap.data(), (size_t)n * 8);
  } else {
    const int32_t nlist = ix->desc.nlist;
    std::vector<float> cents((size_t)nlist * d);
    if (hipMemcpy2D(cents.data(), (size_t)d * 4, ix->d_centroids.p,
                    (size_t)ix->desc.d * 4, (size_t)d * 4, (size_t)nlist,
                    hipMemcpyDeviceToHost) != hipSuccess) {
      fclose(f);
      return DG_EINTERNAL;
    }
    std::vector<std::vector<int64_t>> lists;
    int64_t kept = 0;
    group_rows(h, nlist, lists, &kept);
    w.u32(is_pq ? fcc("IwPQ") : fcc("IwFl"));
    // ivf header: index header, nlist, nprobe, quantizer, direct map
    write_index_header(w, d, kept, ix->trained, fm);
    w.u64((uint64_t)nlist);
    w.u64(1);  // nprobe member default (searches pass IVFSearchParameters)
    write_flat_index(w, d, nlist, fm, cents.data());
    write_direct_map(w);
    if (is_pq) {
      const int32_t M = ix->desc.pq_m;
      const int32_t dsub = d / M;
      std::vector<float> cb((size_t)M * 256 * dsub);
      if (hipMemcpy(cb.data(), ix->d_codebooks.p, cb.size() * 4,
                    hipMemcpyDeviceToHost) != hipSuccess) {
        fclose(f);
        return DG_EINTERNAL;
      }
      w.u8(1);          // by_residual (bool; faiss >= 1.7 writes 1 byte)
      w.u64((uint64_t)M);  // code_size (M codes x 1 byte at nbits=8)
      w.u64((uint64_t)d);  // ProductQuantizer.d
      w.u64((uint64_t)M);  // .M
      w.u64(8);            // .nbits
      w.u64(cb.size());    // centroids vector<float>
      w.raw(cb.data(), cb.size() * 4);
      write_invlists(w, h, lists, M, d, true, M);
    } else {
      write_invlists(w, h, lists, (size_t)d * 4, d, false, 0);
    }
  }
  bool ok = w.ok && fclose(f) == 0;
  if (!ok) {
    dg_set_error("write failed for %s", path);
    return DG_EIO;
  }
  return DG_OK;
}

static dg_status load_faiss_impl(dg_index** out, const char* path,
                                 int32_t metric_override, int32_t device);

extern "C" dg_status dg_load_faiss(dg_index** out, const char* path,
                                   int32_t metric_override, int32_t device) {
  // exception wall: a malformed container can drive a vector resize into
  // std::bad_alloc; no exception may cross the C ABI
  try {
    return load_faiss_impl(out, path, metric_override, device);
  } catch (const std::exception& e) {
    dg_set_error("faiss container load failed: %s", e.what());
    return DG_EIO;
  } catch (...) {
    dg_set_error("faiss container load failed");
    return DG_EIO;
  }
}

static dg_status load_faiss_impl(dg_index** out, const char* path,
                                 int32_t metric_override, int32_t device) {
  if (!out || !path) return DG_EINVAL;
  FILE* f = fopen(path, "rb");
  if (!f) {
    dg_set_error("cannot open %s", path);
    return DG_EIO;
  }
  FReader r{f};
  dg_status st = DG_EIO;
  dg_index* ix = nullptr;
  do {
    uint32_t h4 = r.u32();
    if (!r.ok) break;
    if (h4 == fcc("IxM2") || h4 == fcc("IxMp")) {
      // FLAT: IDMap(2) over IndexFlat
      IdxHeader oh;
      if (!read_index_header(r, oh)) break;
      IdxHeader ih;
      std::vector<float> data;
      if (!read_flat_index(r, ih, data)) break;
      uint64_t nmap = r.u64();
      if (!r.ok || (int64_t)nmap != ih.ntotal) break;
      std::vector<int64_t> idmap(nmap);
      if (nmap) r.raw(idmap.data(), nmap * 8);
      if (!r.ok) break;
      int32_t metric = metric_override >= 0
                           ? metric_override
                           : (ih.metric == 1 ? DG_METRIC_L2 : DG_METRIC_IP);
      if ((ih.metric == 1) != (metric == DG_METRIC_L2)) {
        dg_set_error("metric override incompatible with file metric");
        st = DG_EINVAL;
        break;
      }
      dg_index_desc desc{};
      desc.kind = DG_INDEX_FLAT;
      desc.metric = metric;
      desc.d = ih.d;
      desc.device = device;
      desc.reserve = ih.ntotal;
      if ((st = dg_index_create(&ix, &desc)) != DG_OK) break;
      st = nmap ? dg_ingest_rows(ix, nmap, idmap.data(), data.data(),
                                 nullptr, nullptr)
                : DG_OK;
    } else if (h4 == fcc("IwFl") || h4 == fcc("IwPQ")) {
      IdxHeader oh;
      if (!read_index_header(r, oh)) break;
      uint64_t nlist = r.u64();
      (void)r.u64();  // nprobe member (per-search in the reference)
      IdxHeader qh;
      std::vector<float> cents;
      if (!read_flat_index(r, qh, cents)) break;
      if ((uint64_t)qh.ntotal != nlist || qh.d != oh.d) break;
      if (!read_direct_map(r)) break;
      int32_t metric = metric_override >= 0
                           ? metric_override
                           : (oh.metric == 1 ? DG_METRIC_L2 : DG_METRIC_IP);
      if ((oh.metric == 1) != (metric == DG_METRIC_L2)) {
        dg_set_error("metric override incompatible with file metric");
        st = DG_EINVAL;
        break;
      }
      dg_index_desc desc{};
      desc.metric = metric;
      desc.d = oh.d;
      desc.nlist = (int32_t)nlist;
      desc.device = device;
      desc.reserve = oh.ntotal;
      std::vector<float> cb;
      int32_t M = 0;
      if (h4 == fcc("IwPQ")) {
        desc.kind = DG_INDEX_IVF_PQ;
        uint8_t by_residual = r.u8();
        uint64_t code_size = r.u64();
        if (!by_residual) {
          // non-residual IVFPQ files would decode with the wrong ADC
          dg_set_error("IwPQ with by_residual=0 unsupported");
          st = DG_ENOT_SUPPORT;
          break;
        }
        uint64_t pq_d = r.u64();
        uint64_t pq_M = r.u64();
        uint64_t pq_nbits = r.u64();
        if (!r.ok || pq_nbits != 8 || pq_d != (uint64_t)oh.d ||
            pq_M != code_size || pq_M == 0 || oh.d % pq_M != 0) {
          dg_set_error(
              "unsupported IwPQ parameters (code_size=%llu M=%llu nbits=%llu"
              " — only byte codes with by_residual are supported)",
              (unsigned long long)code_size, (unsigned long long)pq_M,
              (unsigned long long)pq_nbits);
          st = DG_ENOT_SUPPORT;
          break;
        }
        M = (int32_t)pq_M;
        desc.pq_m = M;
        desc.pq_nbits = 8;
        uint64_t nc = r.u64();
        if (!r.ok || nc != (uint64_t)M * 256 * (oh.d / M)) break;
        cb.resize(nc);
        r.raw(cb.data(), nc * 4);
        if (!r.ok) break;
      } else {
        desc.kind = DG_INDEX_IVF_FLAT;
      }
      InvLists il;
      if (!read_invlists(r, il)) break;
      if (il.nlist != nlist) break;
      size_t want_cs = h4 == fcc("IwPQ") ? (size_t)M : (size_t)oh.d * 4;
      if (il.code_size != want_cs) break;
      if ((st = dg_index_create(&ix, &desc)) != DG_OK) break;
      if ((st = dg_set_centroids(ix, (int32_t)nlist, cents.data())) != DG_OK)
        break;
      if (h4 == fcc("IwPQ") &&
          (st = dg_set_codebooks(ix, M, 8, cb.data())) != DG_OK)
        break;
      // flatten lists into arrival arrays with their file assignment
      std::vector<int64_t> ids;
      std::vector<int32_t> assign;
      std::vector<float> vecs;
      std::vector<uint8_t> codes;
      for (uint64_t l = 0; l < nlist; l++) {
        size_t n = il.ids[l].size();
        if (!n) continue;
        ids.insert(ids.end(), il.ids[l].begin(), il.ids[l].end());
        assign.insert(assign.end(), n, (int32_t)l);
        if (h4 == fcc("IwPQ"))
          codes.insert(codes.end(), il.codes[l].begin(), il.codes[l].end());
        else {
          const float* v = (const float*)il.codes[l].data();
          vecs.insert(vecs.end(), v, v + n * oh.d);
        }
      }
      st = ids.empty()
               ? DG_OK
               : dg_ingest_rows(ix, (int64_t)ids.size(), ids.data(),
                                vecs.empty() ? nullptr : vecs.data(),
                                codes.empty() ? nullptr : codes.data(),
                                assign.data());
    } else {
      dg_set_error("unsupported faiss index fourcc 0x%08x", h4);
      st = DG_ENOT_SUPPORT;
    }
  } while (0);
  fclose(f);
  if (st != DG_OK) {
    if (ix) dg_index_destroy(ix);
    if (st == DG_EIO) dg_set_error("malformed faiss container %s", path);
    return st;
  }
  *out = ix;
  return DG_OK;
}
