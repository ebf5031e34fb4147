"""pyoracle — ctypes wrapper over oracle/liboracle.so.

TEST INFRASTRUCTURE ONLY (see oracle/oracle.h): importable only from
tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg.
"""
import ctypes as C
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))

L2, IP, COSINE = 0, 1, 2

_f32p = np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS")
_i64p = np.ctypeslib.ndpointer(np.int64, flags="C_CONTIGUOUS")
_i32p = np.ctypeslib.ndpointer(np.int32, flags="C_CONTIGUOUS")
_u8p = np.ctypeslib.ndpointer(np.uint8, flags="C_CONTIGUOUS")


def _load():
    lib = C.CDLL(os.path.join(_DIR, "liboracle.so"))
    lib.dgo_version.restype = C.c_char_p
    lib.dgo_fvec_L2sqr.restype = C.c_float
    lib.dgo_fvec_L2sqr.argtypes = [_f32p, _f32p, C.c_size_t]
    lib.dgo_fvec_inner_product.restype = C.c_float
    lib.dgo_fvec_inner_product.argtypes = [_f32p, _f32p, C.c_size_t]
    lib.dgo_fvec_norm_L2sqr.restype = C.c_float
    lib.dgo_fvec_norm_L2sqr.argtypes = [_f32p, C.c_size_t]
    lib.dgo_normalize_batch.argtypes = [_f32p, C.c_int64, C.c_int32]
    return lib


_lib = _load()


def version():
    return _lib.dgo_version().decode()


def normalize(x):
    x = np.ascontiguousarray(x, np.float32)
    _lib.dgo_normalize_batch(x, x.shape[0], x.shape[1])
    return x


def fvec_l2sqr(x, y):
    return _lib.dgo_fvec_L2sqr(x, y, x.shape[0])


def fvec_ip(x, y):
    return _lib.dgo_fvec_inner_product(x, y, x.shape[0])


def fvec_norm(x):
    return _lib.dgo_fvec_norm_L2sqr(x, x.shape[0])


def flat_search(metric, base, queries, k, ids=None, fast=False):
    base = np.ascontiguousarray(base, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    n, d = base.shape
    nq = queries.shape[0]
    out_dist = np.empty((nq, k), np.float32)
    out_ids = np.empty((nq, k), np.int64)
    fn = _lib.dgo_flat_search_fast if fast else _lib.dgo_flat_search
    idp = (
        np.ascontiguousarray(ids, np.int64).ctypes.data_as(C.c_void_p)
        if ids is not None
        else None
    )
    fn.argtypes = [
        C.c_int, C.c_int64, C.c_int32, _f32p, C.c_void_p, C.c_int64, _f32p,
        C.c_int32, _f32p, _i64p,
    ]
    fn(metric, n, d, base, idp, nq, queries, k, out_dist, out_ids)
    return out_dist, out_ids


def kmeans(metric, x, nlist, niter=25, seed=1234):
    x = np.ascontiguousarray(x, np.float32)
    n, d = x.shape
    cents = np.empty((nlist, d), np.float32)
    _lib.dgo_kmeans.argtypes = [
        C.c_int, C.c_int64, C.c_int32, _f32p, C.c_int32, C.c_int32,
        C.c_uint32, _f32p,
    ]
    _lib.dgo_kmeans(metric, n, d, x, nlist, niter, seed, cents)
    return cents


def ivf_assign(metric, x, centroids):
    x = np.ascontiguousarray(x, np.float32)
    centroids = np.ascontiguousarray(centroids, np.float32)
    n, d = x.shape
    out = np.empty(n, np.int32)
    _lib.dgo_ivf_assign.argtypes = [
        C.c_int, C.c_int64, C.c_int32, _f32p, C.c_int32, _f32p, _i32p,
    ]
    _lib.dgo_ivf_assign(metric, n, d, x, centroids.shape[0], centroids, out)
    return out


def ivf_build(x, ids, nlist, assign):
    x = np.ascontiguousarray(x, np.float32)
    n, d = x.shape
    offsets = np.empty(nlist + 1, np.int64)
    gv = np.empty_like(x)
    gi = np.empty(n, np.int64)
    ids = (
        np.ascontiguousarray(ids, np.int64)
        if ids is not None
        else np.arange(n, dtype=np.int64)
    )
    _lib.dgo_ivf_build.argtypes = [
        C.c_int64, C.c_int32, _f32p, _i64p, C.c_int32, _i32p, _i64p, _f32p,
        _i64p,
    ]
    _lib.dgo_ivf_build(n, d, x, ids, nlist, np.ascontiguousarray(assign, np.int32), offsets, gv, gi)
    return offsets, gv, gi


def coarse_probe(metric, centroids, queries, nprobe):
    centroids = np.ascontiguousarray(centroids, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    nlist, d = centroids.shape
    nq = queries.shape[0]
    out = np.empty((nq, nprobe), np.int32)
    _lib.dgo_coarse_probe.argtypes = [
        C.c_int, C.c_int32, C.c_int32, _f32p, C.c_int64, _f32p, C.c_int32,
        _i32p,
    ]
    _lib.dgo_coarse_probe(metric, nlist, d, centroids, nq, queries, nprobe, out)
    return out


def ivf_search(metric, centroids, offsets, gv, gi, queries, k, nprobe,
               list_mask=None, fast=False):
    centroids = np.ascontiguousarray(centroids, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    nlist, d = centroids.shape
    nq = queries.shape[0]
    out_dist = np.empty((nq, k), np.float32)
    out_ids = np.empty((nq, k), np.int64)
    fn = _lib.dgo_ivf_search_fast if fast else _lib.dgo_ivf_search
    maskp = (
        np.ascontiguousarray(list_mask, np.uint8).ctypes.data_as(C.c_void_p)
        if list_mask is not None
        else None
    )
    fn.argtypes = [
        C.c_int, C.c_int32, C.c_int32, _f32p, _i64p, _f32p, _i64p, C.c_int64,
        _f32p, C.c_int32, C.c_int32, C.c_void_p, _f32p, _i64p,
    ]
    fn(metric, nlist, d, centroids, offsets, gv, gi, nq, queries, k, nprobe,
       maskp, out_dist, out_ids)
    return out_dist, out_ids


def ivf_search_indexed_fast(metric, centroids, member_offsets, member_rows,
                            base, queries, k, nprobe):
    """CPU-baseline leg: indexed scan over the UNGROUPED base array."""
    centroids = np.ascontiguousarray(centroids, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    base = np.ascontiguousarray(base, np.float32)
    nlist, d = centroids.shape
    nq = queries.shape[0]
    out_dist = np.empty((nq, k), np.float32)
    out_ids = np.empty((nq, k), np.int64)
    fn = _lib.dgo_ivf_search_indexed_fast
    fn.argtypes = [
        C.c_int, C.c_int32, C.c_int32, _f32p, _i64p, _i64p, _f32p, C.c_int64,
        _f32p, C.c_int32, C.c_int32, _f32p, _i64p,
    ]
    fn(metric, nlist, d, centroids,
       np.ascontiguousarray(member_offsets, np.int64),
       np.ascontiguousarray(member_rows, np.int64), base, nq, queries, k,
       nprobe, out_dist, out_ids)
    return out_dist, out_ids


def flat_range_search(metric, base, queries, radius, ids=None, cap=1 << 22):
    base = np.ascontiguousarray(base, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    n, d = base.shape
    nq = queries.shape[0]
    lims = np.zeros(nq + 1, np.int64)
    out_dist = np.empty(cap, np.float32)
    out_ids = np.empty(cap, np.int64)
    fn = _lib.dgo_flat_range_search
    idp = (np.ascontiguousarray(ids, np.int64).ctypes.data_as(C.c_void_p)
           if ids is not None else None)
    fn.argtypes = [C.c_int, C.c_int64, C.c_int32, _f32p, C.c_void_p,
                   C.c_int64, _f32p, C.c_float, _i64p, C.c_int64, _f32p,
                   _i64p]
    fn(metric, n, d, base, idp, nq, queries, radius, lims, cap, out_dist,
       out_ids)
    t = int(lims[-1])
    return lims, out_dist[:t].copy(), out_ids[:t].copy()


def ivf_range_search(metric, centroids, offsets, gv, gi, queries, radius,
                     nprobe, cap=1 << 22):
    centroids = np.ascontiguousarray(centroids, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    nlist, d = centroids.shape
    nq = queries.shape[0]
    lims = np.zeros(nq + 1, np.int64)
    out_dist = np.empty(cap, np.float32)
    out_ids = np.empty(cap, np.int64)
    fn = _lib.dgo_ivf_range_search
    fn.argtypes = [C.c_int, C.c_int32, C.c_int32, _f32p, _i64p, _f32p,
                   _i64p, C.c_int64, _f32p, C.c_float, C.c_int32, _i64p,
                   C.c_int64, _f32p, _i64p]
    fn(metric, nlist, d, centroids, offsets, gv, gi, nq, queries, radius,
       nprobe, lims, cap, out_dist, out_ids)
    t = int(lims[-1])
    return lims, out_dist[:t].copy(), out_ids[:t].copy()


def pq_train(residuals, m, nbits=8, seed=1234):
    residuals = np.ascontiguousarray(residuals, np.float32)
    n, d = residuals.shape
    ksub, dsub = 1 << nbits, d // m
    cb = np.empty((m, ksub, dsub), np.float32)
    _lib.dgo_pq_train.argtypes = [
        C.c_int64, C.c_int32, _f32p, C.c_int32, C.c_int32, C.c_uint32, _f32p,
    ]
    _lib.dgo_pq_train(n, d, residuals, m, nbits, seed, cb.reshape(-1))
    return cb


def ivfpq_encode(x, assign, centroids, codebooks):
    x = np.ascontiguousarray(x, np.float32)
    n, d = x.shape
    m = codebooks.shape[0]
    codes = np.empty((n, m), np.uint8)
    _lib.dgo_ivfpq_encode.argtypes = [
        C.c_int64, C.c_int32, _f32p, _i32p, _f32p, C.c_int32, _f32p, _u8p,
    ]
    _lib.dgo_ivfpq_encode(
        n, d, x, np.ascontiguousarray(assign, np.int32),
        np.ascontiguousarray(centroids, np.float32), m,
        np.ascontiguousarray(codebooks, np.float32).reshape(-1), codes)
    return codes


def ivfpq_search(metric, centroids, offsets, gcodes, gi, codebooks, queries,
                 k, nprobe):
    centroids = np.ascontiguousarray(centroids, np.float32)
    queries = np.ascontiguousarray(queries, np.float32)
    nlist, d = centroids.shape
    m = codebooks.shape[0]
    nq = queries.shape[0]
    out_dist = np.empty((nq, k), np.float32)
    out_ids = np.empty((nq, k), np.int64)
    _lib.dgo_ivfpq_search.argtypes = [
        C.c_int, C.c_int32, C.c_int32, _f32p, _i64p, _u8p, _i64p, C.c_int32,
        _f32p, C.c_int64, _f32p, C.c_int32, C.c_int32, _f32p, _i64p,
    ]
    _lib.dgo_ivfpq_search(
        metric, nlist, d, centroids, offsets,
        np.ascontiguousarray(gcodes, np.uint8), gi, m,
        np.ascontiguousarray(codebooks, np.float32).reshape(-1), nq, queries,
        k, nprobe, out_dist, out_ids)
    return out_dist, out_ids
