"""Oracle validation (CPU, no GPU needed).

Pins, in order of strength:
1. BIT-EXACT vs the reference's own in-tree scalar arithmetic
   (src/simd/distances_ref.cc compiled unmodified into oracle/_ref).
2. Tolerance vs numpy float64 recomputation.
3. The behavioral pins the reference's own test suite holds
   (Flat self-top-1, test_vector_index_recall_flat.cc:170-236).
4. Size-independent properties (flat == ivf at nprobe=nlist; shard-union
   invariance; recall vs exhaustive ground truth).
5. Committed golden fixtures (tests/golden/) guarding regressions.
"""
import ctypes as C
import os

import numpy as np
import pytest

import pyoracle as orc
import workload

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = os.path.join(HERE, "golden")
REF_SO = os.path.join(HERE, "..", "oracle", "_ref", "libdistref.so")

rng = np.random.default_rng(7)


def _rand(n, d, scale=1.0):
    return (rng.random((n, d), dtype=np.float32) * scale).astype(np.float32)


# ---------- 1. bit-exact vs reference scalar code ----------
@pytest.mark.skipif(not os.path.exists(REF_SO), reason="_ref not built")
def test_core_bitexact_vs_reference():
    ref = C.CDLL(REF_SO)
    fp = np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS")
    for name in ("ref_fvec_L2sqr", "ref_fvec_inner_product"):
        getattr(ref, name).restype = C.c_float
        getattr(ref, name).argtypes = [fp, fp, C.c_size_t]
    ref.ref_fvec_norm_L2sqr.restype = C.c_float
    ref.ref_fvec_norm_L2sqr.argtypes = [fp, C.c_size_t]

    for d in (1, 3, 64, 128, 768, 1000):
        x = _rand(1, d)[0] * 3 - 1
        y = _rand(1, d)[0] * 3 - 1
        assert orc.fvec_l2sqr(x, y) == ref.ref_fvec_L2sqr(x, y, d)
        assert orc.fvec_ip(x, y) == ref.ref_fvec_inner_product(x, y, d)
        assert orc.fvec_norm(x) == ref.ref_fvec_norm_L2sqr(x, d)


# ---------- 2. tolerance vs float64 ----------
def test_core_vs_float64():
    for d in (8, 768):
        x = _rand(1, d)[0]
        y = _rand(1, d)[0]
        l2_64 = float(np.sum((x.astype(np.float64) - y.astype(np.float64)) ** 2))
        ip_64 = float(np.dot(x.astype(np.float64), y.astype(np.float64)))
        assert abs(orc.fvec_l2sqr(x, y) - l2_64) <= 1e-4 * max(1.0, abs(l2_64))
        assert abs(orc.fvec_ip(x, y) - ip_64) <= 1e-4 * max(1.0, abs(ip_64))


def test_normalize_semantics():
    # NormalizeVectorForFaiss: skips when |1 - ||x||^2| <= 1e-5
    x = np.zeros((1, 4), np.float32)
    x[0, 0] = 1.0 + 3e-6
    before = x.copy()
    orc.normalize(x)
    assert np.array_equal(x, before)  # within accuracy window: untouched
    y = np.full((1, 4), 2.0, np.float32)
    orc.normalize(y)
    assert abs(orc.fvec_norm(y[0]) - 1.0) < 1e-5
    z = np.zeros((1, 4), np.float32)  # zero vector untouched
    orc.normalize(z)
    assert np.array_equal(z, np.zeros((1, 4), np.float32))


# ---------- 3. reference behavioral pin: flat self-top-1 ----------
def test_flat_self_top1():
    base = _rand(500, 32)
    for metric in (orc.L2, orc.COSINE):
        b = base.copy()
        if metric == orc.COSINE:
            b = orc.normalize(b)
        dist, ids = orc.flat_search(metric, b, b, 1)
        assert np.array_equal(ids[:, 0], np.arange(500)), f"metric {metric}"


def test_flat_matches_numpy_f64():
    base = _rand(2000, 48)
    q = _rand(64, 48)
    k = 10
    for metric in (orc.L2, orc.IP):
        dist, ids = orc.flat_search(metric, base, q, k)
        if metric == orc.L2:
            full = ((q[:, None, :].astype(np.float64) -
                     base[None, :, :].astype(np.float64)) ** 2).sum(-1)
            ref_ids = np.argsort(full, axis=1, kind="stable")[:, :k]
        else:
            full = -q.astype(np.float64) @ base.T.astype(np.float64)
            ref_ids = np.argsort(full, axis=1, kind="stable")[:, :k]
        assert (ids == ref_ids).mean() > 0.999  # fp-tie slack
        # distances agree with f64 within tolerance
        d64 = np.take_along_axis(full, ids, 1)
        if metric == orc.IP:
            d64 = -d64
        assert np.allclose(dist, d64, rtol=1e-4, atol=1e-4)


def test_flat_explicit_ids_and_padding():
    base = _rand(5, 16)
    ids = np.array([100, 50, 7, 900, 3], np.int64)
    dist, out = orc.flat_search(orc.L2, base, base[:2], 8, ids=ids)
    assert set(out[0, :5]) == set(ids)
    assert (out[:, 5:] == -1).all()
    assert out[0, 0] == 100  # self is top-1


# ---------- 4. IVF properties ----------
def _build_ivf(metric, base, nlist, seed=1234):
    cents = orc.kmeans(metric, workloads_sub(base), nlist, seed=seed)
    assign = orc.ivf_assign(metric, base, cents)
    offsets, gv, gi = orc.ivf_build(base, None, nlist, assign)
    return cents, offsets, gv, gi


def workloads_sub(base):
    return base  # small tests: train on everything


def test_ivf_nprobe_all_equals_flat():
    base = _rand(3000, 32)
    q = _rand(40, 32)
    nlist, k = 16, 10
    for metric in (orc.L2, orc.IP):
        cents, offsets, gv, gi = _build_ivf(metric, base, nlist)
        fd, fi = orc.flat_search(metric, base, q, k)
        vd, vi = orc.ivf_search(metric, cents, offsets, gv, gi, q, k, nlist)
        assert np.array_equal(fi, vi)
        assert np.allclose(fd, vd, rtol=1e-5, atol=1e-5)


def test_ivf_equals_exact_scan_of_probed_lists():
    """THE IVF correctness pin: the IVF result is exactly a flat search
    restricted to the members of the query's probed lists (faiss
    IndexIVFFlat semantics — raw vectors, no further approximation)."""
    n, d, nlist, nprobe, k = 20000, 64, 64, 8, 10
    base = workload.gen_base(4242, n, d)
    q = workload.gen_queries(4242, n, d, 200)
    cents, offsets, gv, gi = _build_ivf(orc.L2, base, nlist)
    vd, vi = orc.ivf_search(orc.L2, cents, offsets, gv, gi, q, k, nprobe)
    probes = orc.coarse_probe(orc.L2, cents, q, nprobe)
    for qi in range(0, q.shape[0], 17):  # sample queries
        member_rows = np.concatenate([
            np.arange(offsets[l], offsets[l + 1]) for l in probes[qi]])
        sub = gv[member_rows]
        sub_ids = gi[member_rows]
        sd, si = orc.flat_search(orc.L2, sub, q[qi:qi + 1], k,
                                 ids=sub_ids)
        assert np.array_equal(si[0], vi[qi])
        assert np.allclose(sd[0], vd[qi], rtol=1e-5, atol=1e-5)
    # recall vs exhaustive ground truth: reported, loosely sanity-checked
    gt_d, gt_i = orc.flat_search(orc.L2, base, q, k)
    recall = np.mean([len(set(a) & set(b)) / k for a, b in zip(gt_i, vi)])
    assert recall > 0.25, recall  # uniform data, 1/8 of lists probed


def test_ivf_shard_union_invariance():
    """Row-sharded search union == whole search (multi-GPU §8e invariant)."""
    n, d, nlist, nprobe, k = 6000, 32, 32, 8, 10
    base = workload.gen_base(99, n, d)
    q = workload.gen_queries(99, n, d, 32)
    cents = orc.kmeans(orc.L2, base, nlist)
    # whole
    assign = orc.ivf_assign(orc.L2, base, cents)
    off, gv, gi = orc.ivf_build(base, None, nlist, assign)
    wd, wi = orc.ivf_search(orc.L2, cents, off, gv, gi, q, k, nprobe)
    # two row shards, same centroids
    merged = []
    shard_results = []
    for r in range(2):
        rows = slice(r * n // 2, (r + 1) * n // 2)
        ids = np.arange(rows.start, rows.stop, dtype=np.int64)
        a = orc.ivf_assign(orc.L2, base[rows], cents)
        o2, v2, i2 = orc.ivf_build(base[rows], ids, nlist, a)
        shard_results.append(orc.ivf_search(orc.L2, cents, o2, v2, i2, q, k, nprobe))
    # merge k best by (dist, id)
    for qi in range(q.shape[0]):
        cand = []
        for sd, si in shard_results:
            for j in range(k):
                if si[qi, j] >= 0:
                    cand.append((sd[qi, j], si[qi, j]))
        cand.sort()
        merged.append([c[1] for c in cand[:k]])
    assert np.array_equal(np.array(merged), wi)


def test_ivf_list_mask():
    n, d, nlist = 2000, 16, 8
    base = _rand(n, d)
    q = _rand(8, d)
    cents, off, gv, gi = _build_ivf(orc.L2, base, nlist)
    mask = np.zeros(nlist, np.uint8)
    mask[: nlist // 2] = 1
    vd, vi = orc.ivf_search(orc.L2, cents, off, gv, gi, q, 5, nlist,
                            list_mask=mask)
    # every returned id must live in a masked list
    ok_ids = set()
    for l in range(nlist // 2):
        ok_ids.update(gi[off[l]:off[l + 1]])
    for row in vi:
        for i in row:
            assert i == -1 or i in ok_ids


def test_fast_matches_strict():
    base = _rand(4000, 96)
    q = _rand(64, 96)
    k = 10
    sd, si = orc.flat_search(orc.L2, base, q, k)
    fd, fi = orc.flat_search(orc.L2, base, q, k, fast=True)
    assert (si == fi).mean() > 0.995
    assert np.allclose(sd, fd, rtol=1e-4, atol=1e-4)


def test_kmeans_deterministic_and_converges():
    x = _rand(5000, 24)
    c1 = orc.kmeans(orc.L2, x, 16)
    c2 = orc.kmeans(orc.L2, x, 16)
    assert np.array_equal(c1, c2)
    # objective no worse than 1-iter clustering
    c0 = orc.kmeans(orc.L2, x, 16, niter=1)
    def obj(c):
        a = orc.ivf_assign(orc.L2, x, c)
        return float(np.sum((x - c[a]) ** 2))
    assert obj(c1) <= obj(c0) * 1.001


# ---------- IVF-PQ ----------
def test_ivfpq_recall_sane():
    n, d, m, nlist, nprobe, k = 8000, 64, 8, 16, 8, 10
    base = workload.gen_base(11, n, d)
    q = workload.gen_queries(11, n, d, 64)
    cents = orc.kmeans(orc.L2, base, nlist)
    assign = orc.ivf_assign(orc.L2, base, cents)
    residuals = base - cents[assign]
    cb = orc.pq_train(residuals, m)
    codes = orc.ivfpq_encode(base, assign, cents, cb)
    off, _, gi = orc.ivf_build(base, None, nlist, assign)
    gcodes = np.empty_like(codes)
    # group codes in CSR order (same permutation ivf_build applied)
    cursor = off[:-1].copy()
    for i in range(n):
        gcodes[cursor[assign[i]]] = codes[i]
        cursor[assign[i]] += 1
    pd, pi = orc.ivfpq_search(orc.L2, cents, off, gcodes, gi, cb, q, k, nprobe)
    gt_d, gt_i = orc.flat_search(orc.L2, base, q, k)
    recall = np.mean([len(set(a) & set(b)) / k for a, b in zip(gt_i, pi)])
    assert recall > 0.35, recall  # quantized @ m=8: coarse but sane


# ---------- 5. golden fixtures ----------
def test_golden_fixtures():
    path = os.path.join(GOLDEN, "oracle_golden.npz")
    assert os.path.exists(path), "run tests/golden/make_golden.py"
    g = np.load(path)
    base = workload.gen_base(int(g["seed"]), int(g["n"]), int(g["d"]))
    q = workload.gen_queries(int(g["seed"]), int(g["n"]), int(g["d"]),
                             int(g["nq"]))
    fd, fi = orc.flat_search(orc.L2, base, q, int(g["k"]))
    assert np.array_equal(fi, g["flat_ids"])
    assert np.array_equal(fd, g["flat_dist"])  # bit-exact regression pin
    cents = orc.kmeans(orc.L2, base, int(g["nlist"]))
    assert np.array_equal(cents, g["centroids"])
    assign = orc.ivf_assign(orc.L2, base, cents)
    off, gv, gi_ = orc.ivf_build(base, None, int(g["nlist"]), assign)
    vd, vi = orc.ivf_search(orc.L2, cents, off, gv, gi_, q, int(g["k"]),
                            int(g["nprobe"]))
    assert np.array_equal(vi, g["ivf_ids"])
    assert np.array_equal(vd, g["ivf_dist"])


def test_golden_pq_and_range():
    g = np.load(os.path.join(GOLDEN, "oracle_golden.npz"))
    base = workload.gen_base(int(g["seed"]), int(g["n"]), int(g["d"]))
    q = workload.gen_queries(int(g["seed"]), int(g["n"]), int(g["d"]),
                             int(g["nq"]))
    nlist, k, nprobe = int(g["nlist"]), int(g["k"]), int(g["nprobe"])
    cents = g["centroids"]
    assign = orc.ivf_assign(orc.L2, base, cents)
    off, gv, gi_ = orc.ivf_build(base, None, nlist, assign)
    m = int(g["pq_m"])
    cb = orc.pq_train(base - cents[assign], m)
    assert np.array_equal(cb, g["codebooks"])  # deterministic PQ train
    codes = orc.ivfpq_encode(base, assign, cents, cb)
    gcodes = np.empty_like(codes)
    cursor = off[:-1].copy()
    for i in range(int(g["n"])):
        gcodes[cursor[assign[i]]] = codes[i]
        cursor[assign[i]] += 1
    pd, pi = orc.ivfpq_search(orc.L2, cents, off, gcodes, gi_, cb, q, k,
                              nprobe)
    assert np.array_equal(pi, g["pq_ids"])
    assert np.array_equal(pd, g["pq_dist"])
    rl, rd, ri = orc.flat_range_search(orc.L2, base, q,
                                       float(g["range_radius"]))
    assert np.array_equal(rl, g["range_lims"])
    assert np.array_equal(ri, g["range_ids"])
    assert np.array_equal(rd, g["range_dist"])


def test_bruteforce_batching_equals_whole_scan():
    """Reader brute-force semantics (vector_reader.cc:1873-2048): scanning
    in 2048-row batches through per-batch exact Flat searches and merging
    per-query top-k by distance (max-heap keeps the smallest topk; output
    ascending) must equal one exact search over all rows.  This is the CPU
    statement of the property dg_mirror_selftest pins on the GPU mirror."""
    import heapq
    rng = np.random.default_rng(99)
    n, d, nq, k, batch = 5000, 24, 6, 7, 2048
    base = rng.random((n, d), dtype=np.float32)
    ids = np.arange(n, dtype=np.int64) * 2 + 1
    q = base[:nq] + rng.normal(0, 0.05, (nq, d)).astype(np.float32)

    wd, wi = orc.flat_search(orc.L2, base, q, k, ids=ids)

    tops = [[] for _ in range(nq)]  # python heapq = min-heap; negate
    for s0 in range(0, n, batch):
        bd, bi = orc.flat_search(orc.L2, base[s0:s0 + batch], q, k,
                                 ids=ids[s0:s0 + batch])
        for r in range(nq):
            for dist, vid in zip(bd[r], bi[r]):
                if vid < 0:
                    continue
                if len(tops[r]) < k:
                    heapq.heappush(tops[r], (-dist, -vid))
                elif -tops[r][0][0] > dist:  # current worst > new
                    heapq.heapreplace(tops[r], (-dist, -vid))
    for r in range(nq):
        merged = sorted(((-md, -mv) for md, mv in tops[r]))
        np.testing.assert_array_equal([v for _, v in merged], wi[r])
        np.testing.assert_allclose([dd for dd, _ in merged], wd[r],
                                   rtol=1e-6, atol=1e-6)
