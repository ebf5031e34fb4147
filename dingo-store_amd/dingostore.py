"""dingostore — ctypes wrapper over the MI355X-native index library
(dingo-store_amd/libdingo_gpu.so).

This is the PRODUCT path: it loads the HIP library built for gfx950 and
fails loudly if the library or a GPU is missing.  There is no CPU fallback
anywhere behind these calls.
"""
import ctypes as C
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libdingo_gpu.so")

L2, IP, COSINE = 0, 1, 2
FLAT, IVF_FLAT, IVF_PQ = 0, 1, 2

_f32p = np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS")
_i64p = np.ctypeslib.ndpointer(np.int64, flags="C_CONTIGUOUS")


class DgError(RuntimeError):
    pass


class _Desc(C.Structure):
    _fields_ = [
        ("kind", C.c_int32), ("metric", C.c_int32), ("d", C.c_int32),
        ("nlist", C.c_int32), ("pq_m", C.c_int32), ("pq_nbits", C.c_int32),
        ("device", C.c_int32), ("reserve", C.c_int64),
    ]


class _Filter(C.Structure):
    _fields_ = [
        ("kind", C.c_int32), ("negate", C.c_int32),
        ("min_id", C.c_int64), ("max_id", C.c_int64),
        ("ids", C.c_void_p), ("n_ids", C.c_int64),
        ("bitmap", C.c_void_p), ("bitmap_base", C.c_int64),
        ("bitmap_nbits", C.c_int64),
    ]


class _Stats(C.Structure):
    _fields_ = [
        ("ntotal", C.c_int64), ("d", C.c_int32), ("metric", C.c_int32),
        ("kind", C.c_int32), ("nlist", C.c_int32), ("is_trained", C.c_int32),
        ("_pad", C.c_int32), ("device_bytes", C.c_int64),
        ("last_coarse_ms", C.c_double), ("last_scan_ms", C.c_double),
        ("last_select_ms", C.c_double), ("last_total_ms", C.c_double),
        ("last_nq", C.c_int64),
        ("last_scan_bytes_algorithmic", C.c_int64),
        ("last_scan_gbps_algorithmic", C.c_double),
        ("deleted_count", C.c_int64),
    ]


def _load():
    if not os.path.exists(_SO):
        raise DgError(
            f"{_SO} missing — build it with __graft_entry__.build() "
            "(the GPU product path has no fallback)")
    lib = C.CDLL(_SO)
    lib.dg_build_info.restype = C.c_char_p
    lib.dg_device_count.restype = C.c_int
    lib.dg_last_error.argtypes = [C.c_char_p, C.c_int64]
    lib.dg_index_create.argtypes = [C.POINTER(C.c_void_p), C.POINTER(_Desc)]
    lib.dg_index_destroy.argtypes = [C.c_void_p]
    lib.dg_train.argtypes = [C.c_void_p, C.c_int64, _f32p]
    lib.dg_set_centroids.argtypes = [C.c_void_p, C.c_int32, _f32p]
    lib.dg_get_centroids.argtypes = [C.c_void_p, _f32p]
    lib.dg_add.argtypes = [C.c_void_p, C.c_int64, _i64p, _f32p]
    lib.dg_upsert.argtypes = [C.c_void_p, C.c_int64, _i64p, _f32p]
    lib.dg_remove.argtypes = [C.c_void_p, C.c_int64, _i64p]
    lib.dg_search.argtypes = [
        C.c_void_p, C.c_int64, _f32p, C.c_int32, C.c_int32,
        C.POINTER(_Filter), _f32p, _i64p,
    ]
    lib.dg_search_device.argtypes = [
        C.c_void_p, C.c_int64, C.c_void_p, C.c_int32, C.c_int32,
        C.POINTER(_Filter), C.c_void_p, C.c_void_p,
    ]
    lib.dg_sync.argtypes = [C.c_void_p]
    lib.dg_save.argtypes = [C.c_void_p, C.c_char_p]
    lib.dg_load.argtypes = [C.POINTER(C.c_void_p), C.c_char_p, C.c_int32]
    lib.dg_stats.argtypes = [C.c_void_p, C.POINTER(_Stats)]
    lib.dg_set_list_mask.argtypes = [C.c_void_p, C.c_void_p]
    lib.dg_mirror_selftest.restype = C.c_int
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


def last_error():
    buf = C.create_string_buffer(1024)
    lib().dg_last_error(buf, 1024)
    return buf.value.decode()


def _check(st, what):
    if st != 0:
        raise DgError(f"{what} failed (status {st}): {last_error()}")


def build_info():
    return lib().dg_build_info().decode()


def device_count():
    return lib().dg_device_count()


def make_filter(kind=0, negate=False, min_id=0, max_id=0, ids=None,
                bitmap=None, bitmap_base=0):
    f = _Filter()
    f.kind = kind
    f.negate = 1 if negate else 0
    f.min_id, f.max_id = min_id, max_id
    if ids is not None:
        ids = np.ascontiguousarray(ids, np.int64)
        f._ids_keepalive = ids  # not a ctypes field; python-side keepalive
        f.ids = ids.ctypes.data
        f.n_ids = len(ids)
    if bitmap is not None:
        bitmap = np.ascontiguousarray(bitmap, np.uint64)
        f._bm_keepalive = bitmap
        f.bitmap = bitmap.ctypes.data
        f.bitmap_base = bitmap_base
        f.bitmap_nbits = len(bitmap) * 64
    return f


class Index:
    def __init__(self, kind, metric, d, nlist=0, m=0, device=-1,
                 reserve=0, _handle=None):
        self.kind, self.metric, self.d = kind, metric, d
        self.nlist = nlist
        self.m = m
        if _handle is not None:
            self.h = _handle
            return
        desc = _Desc(kind=kind, metric=metric, d=d, nlist=nlist,
                     pq_m=m, pq_nbits=8, device=device, reserve=reserve)
        h = C.c_void_p()
        _check(lib().dg_index_create(C.byref(h), C.byref(desc)),
               "dg_index_create")
        self.h = h

    def close(self):
        if getattr(self, "h", None):
            lib().dg_index_destroy(self.h)
            self.h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def train(self, x):
        x = np.ascontiguousarray(x, np.float32)
        _check(lib().dg_train(self.h, x.shape[0], x), "dg_train")

    def set_centroids(self, centroids):
        centroids = np.ascontiguousarray(centroids, np.float32)
        self.nlist = centroids.shape[0]
        _check(lib().dg_set_centroids(self.h, centroids.shape[0], centroids),
               "dg_set_centroids")

    def get_centroids(self):
        out = np.empty((self.nlist, self.d), np.float32)
        _check(lib().dg_get_centroids(self.h, out), "dg_get_centroids")
        return out

    def set_codebooks(self, codebooks):
        """codebooks: [m, 256, d//m] float32."""
        cb = np.ascontiguousarray(codebooks, np.float32)
        self.m = cb.shape[0]
        l = lib()
        l.dg_set_codebooks.argtypes = [C.c_void_p, C.c_int32, C.c_int32,
                                       _f32p]
        _check(l.dg_set_codebooks(self.h, cb.shape[0], 8, cb.reshape(-1)),
               "dg_set_codebooks")

    def get_codebooks(self):
        dsub = self.d // self.m
        out = np.empty((self.m, 256, dsub), np.float32)
        l = lib()
        l.dg_get_codebooks.argtypes = [C.c_void_p, _f32p]
        _check(l.dg_get_codebooks(self.h, out.reshape(-1)),
               "dg_get_codebooks")
        return out

    def add(self, ids, x):
        x = np.ascontiguousarray(x, np.float32)
        ids = np.ascontiguousarray(ids, np.int64)
        _check(lib().dg_add(self.h, x.shape[0], ids, x), "dg_add")

    def add_device(self, ids, x_ptr, n):
        """Add n vectors from a device pointer (torch .data_ptr())."""
        ids = np.ascontiguousarray(ids, np.int64)
        l = lib()
        l.dg_add_device.argtypes = [C.c_void_p, C.c_int64, _i64p, C.c_void_p]
        _check(l.dg_add_device(self.h, n, ids, C.c_void_p(x_ptr)),
               "dg_add_device")

    def export_assign(self):
        l = lib()
        l.dg_export_assign.argtypes = [
            C.c_void_p, np.ctypeslib.ndpointer(np.int32,
                                               flags="C_CONTIGUOUS")]
        st = self.stats()
        out = np.empty(st["ntotal"], np.int32)
        _check(l.dg_export_assign(self.h, out), "dg_export_assign")
        return out

    def upsert(self, ids, x):
        x = np.ascontiguousarray(x, np.float32)
        ids = np.ascontiguousarray(ids, np.int64)
        _check(lib().dg_upsert(self.h, x.shape[0], ids, x), "dg_upsert")

    def remove(self, ids):
        ids = np.ascontiguousarray(ids, np.int64)
        _check(lib().dg_remove(self.h, len(ids), ids), "dg_remove")

    def search(self, queries, k, nprobe=0, filt=None):
        q = np.ascontiguousarray(queries, np.float32)
        nq = q.shape[0]
        dist = np.empty((nq, k), np.float32)
        ids = np.empty((nq, k), np.int64)
        fp = C.byref(filt) if filt is not None else None
        _check(lib().dg_search(self.h, nq, q, k, nprobe, fp, dist, ids),
               "dg_search")
        return dist, ids

    def search_device(self, q_ptr, nq, k, nprobe, dist_ptr, ids_ptr,
                      filt=None):
        """Device-pointer hot path (pointers = torch .data_ptr())."""
        fp = C.byref(filt) if filt is not None else None
        _check(lib().dg_search_device(self.h, nq, C.c_void_p(q_ptr), k,
                                      nprobe, fp, C.c_void_p(dist_ptr),
                                      C.c_void_p(ids_ptr)),
               "dg_search_device")

    def range_search(self, queries, radius, filt=None):
        """Radius search (faiss convention: L2 dist < r; IP score > r).
        Returns (lims[nq+1], dists, ids) with per-query best-first order."""
        q = np.ascontiguousarray(queries, np.float32)
        nq = q.shape[0]
        lims = np.zeros(nq + 1, np.int64)
        out_ids = C.POINTER(C.c_int64)()
        out_dists = C.POINTER(C.c_float)()
        l = lib()
        l.dg_range_search.argtypes = [
            C.c_void_p, C.c_int64, _f32p, C.c_float, C.POINTER(_Filter),
            _i64p, C.POINTER(C.POINTER(C.c_int64)),
            C.POINTER(C.POINTER(C.c_float)),
        ]
        l.dg_free.argtypes = [C.c_void_p]
        fp = C.byref(filt) if filt is not None else None
        _check(l.dg_range_search(self.h, nq, q, radius, fp, lims,
                                 C.byref(out_ids), C.byref(out_dists)),
               "dg_range_search")
        total = int(lims[-1])
        dists = np.ctypeslib.as_array(out_dists, (max(total, 1),))[
            :total].copy()
        ids = np.ctypeslib.as_array(out_ids, (max(total, 1),))[:total].copy()
        l.dg_free(C.cast(out_ids, C.c_void_p))
        l.dg_free(C.cast(out_dists, C.c_void_p))
        return lims, dists, ids

    def sync(self):
        _check(lib().dg_sync(self.h), "dg_sync")

    def set_list_mask(self, mask):
        if mask is None:
            _check(lib().dg_set_list_mask(self.h, None), "dg_set_list_mask")
        else:
            mask = np.ascontiguousarray(mask, np.uint8)
            _check(lib().dg_set_list_mask(self.h, mask.ctypes.data),
                   "dg_set_list_mask")

    def save(self, path):
        _check(lib().dg_save(self.h, path.encode()), "dg_save")

    @classmethod
    def load(cls, path, device=-1):
        h = C.c_void_p()
        _check(lib().dg_load(C.byref(h), path.encode(), device), "dg_load")
        return cls._from_handle(h)

    def save_faiss(self, path):
        """Write a faiss-1.7.x-compatible container (the snapshot format the
        reference ships between nodes, vector_index_snapshot_manager.cc)."""
        l = lib()
        l.dg_save_faiss.argtypes = [C.c_void_p, C.c_char_p]
        _check(l.dg_save_faiss(self.h, path.encode()), "dg_save_faiss")

    @classmethod
    def load_faiss(cls, path, metric=-1, device=-1):
        """Load a faiss container (IxM2{IndexFlat} / IwFl / IwPQ).  metric
        = COSINE reinterprets an IP-metric file as a cosine index (the
        reference stores cosine as IP over normalized vectors)."""
        l = lib()
        l.dg_load_faiss.argtypes = [C.POINTER(C.c_void_p), C.c_char_p,
                                    C.c_int32, C.c_int32]
        h = C.c_void_p()
        _check(l.dg_load_faiss(C.byref(h), path.encode(), metric, device),
               "dg_load_faiss")
        return cls._from_handle(h)

    @classmethod
    def _from_handle(cls, h):
        idx = cls.__new__(cls)
        idx.h = h
        st = idx.stats()
        idx.kind, idx.metric = st["kind"], st["metric"]
        idx.d, idx.nlist = st["d"], st["nlist"]
        idx.m = 0
        return idx

    def stats(self):
        s = _Stats()
        _check(lib().dg_stats(self.h, C.byref(s)), "dg_stats")
        return {f[0]: getattr(s, f[0]) for f in _Stats._fields_
                if not f[0].startswith("_")}


def merge_topk(dists, ids, k, metric=L2):
    """Merge per-shard top-k results (the RCCL all-gather consumer).

    dists/ids: arrays [n_shards, nq, k] in faiss convention (L2 raw asc
    best; IP raw score desc best).  Returns merged [nq, k].
    Mirrors the reference's client-side region-scatter merge.
    """
    dists = np.asarray(dists)
    ids = np.asarray(ids)
    S, nq, kk = dists.shape
    cat_d = dists.transpose(1, 0, 2).reshape(nq, S * kk)
    cat_i = ids.transpose(1, 0, 2).reshape(nq, S * kk)
    key = cat_d if metric == L2 else -cat_d
    key = np.where(cat_i < 0, np.inf, key)
    order = np.lexsort((cat_i, key), axis=1)[:, :k]
    return (np.take_along_axis(cat_d, order, 1),
            np.take_along_axis(cat_i, order, 1))
