"""faiss-file-compatible Save/Load.

GPU tests: round-trip every index kind through dg_save_faiss/dg_load_faiss
and pin identical search results; validate the emitted bytes with the
independent pure-Python parser (tests/faiss_format.py, restating the
published faiss 1.7.x container layout the reference's snapshot cycle
ships, vector_index_snapshot_manager.cc:583-599).

CPU test: parse the committed golden container (tests/golden/*.faissindex,
generated on a GPU box by this file's --make-golden hook) and check its
structure + content against the committed expectation arrays.
"""
import os
import sys

import numpy as np
import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
sys.path[:0] = [REPO, HERE, os.path.join(REPO, "oracle"),
                os.path.join(REPO, "dingo-store_amd")]

import faiss_format as ff  # noqa: E402
import workload  # noqa: E402

GOLDEN = os.path.join(HERE, "golden", "ivfflat_small.faissindex")
GOLDEN_NPZ = os.path.join(HERE, "golden", "ivfflat_small_expect.npz")


# ---------------- CPU: committed golden structure ----------------
def test_golden_container_structure():
    if not os.path.exists(GOLDEN):
        pytest.skip("golden faiss container not generated yet")
    exp = np.load(GOLDEN_NPZ)
    idx = ff.read_index(GOLDEN)
    assert idx["kind"] == "ivfflat"
    h = idx["header"]
    assert h["d"] == int(exp["d"]) and h["metric"] == 1  # L2
    assert h["ntotal"] == len(exp["ids"])
    q = idx["quantizer"]
    assert q["fourcc"] == "IxF2" and q["ntotal"] == int(exp["nlist"])
    np.testing.assert_array_equal(q["xb"], exp["centroids"])
    il = idx["invlists"]
    assert il["code_size"] == h["d"] * 4
    # every stored (id, vector) appears exactly once, in its assigned list
    got_ids = np.concatenate(il["ids"])
    assert sorted(got_ids) == sorted(exp["ids"])
    vec_by_id = {int(i): v for i, v in zip(exp["ids"], exp["vectors"])}
    for l in range(int(exp["nlist"])):
        for j, vid in enumerate(il["ids"][l]):
            v = il["codes"][l][j].view(np.float32)
            np.testing.assert_array_equal(v, vec_by_id[int(vid)])
    assert idx["direct_map"]["type"] == 0 and len(idx["direct_map"]["array"]) == 0


def test_golden_container_searches_like_raw_vectors():
    """Semantic pin, CPU-only: an exact search over the vectors as stored
    IN the container must equal the exact search over the original raw
    vectors — i.e. dg_save_faiss wrote the right bytes, not just the right
    structure.  Uses the oracle as the checker."""
    if not os.path.exists(GOLDEN):
        pytest.skip("golden faiss container not generated yet")
    import pyoracle as orc
    exp = np.load(GOLDEN_NPZ)
    idx = ff.read_index(GOLDEN)
    # flatten container rows (vectors + their stored ids)
    vecs = np.concatenate(
        [c.view(np.float32).reshape(-1, idx["header"]["d"])
         if len(c) else np.empty((0, idx["header"]["d"]), np.float32)
         for c in idx["invlists"]["codes"]])
    ids = np.concatenate(idx["invlists"]["ids"]).astype(np.int64)
    q = workload.gen_queries(404, len(exp["ids"]), int(exp["d"]), 16)
    cd, ci = orc.flat_search(orc.L2, vecs, q, 5, ids=ids)
    rd, ri = orc.flat_search(orc.L2, np.asarray(exp["vectors"]), q, 5,
                             ids=np.asarray(exp["ids"], np.int64))
    np.testing.assert_array_equal(ci, ri)
    np.testing.assert_allclose(cd, rd, rtol=1e-6, atol=1e-6)


# ---------------- GPU round trips ----------------
gpu = pytest.mark.gpu


def _dg():
    return pytest.importorskip("dingostore")


@gpu
@pytest.mark.parametrize("metric_name", ["l2", "ip", "cosine"])
def test_flat_roundtrip(tmp_path, metric_name):
    dg = _dg()
    metric = {"l2": dg.L2, "ip": dg.IP, "cosine": dg.COSINE}[metric_name]
    n, d, nq, k = 5000, 64, 32, 10
    base = workload.gen_base(101, n, d)
    q = workload.gen_queries(101, n, d, nq)
    idx = dg.Index(dg.FLAT, metric, d)
    ids = np.arange(n, dtype=np.int64) * 3 + 7
    idx.add(ids, base)
    idx.remove(ids[100:200])  # tombstones must be dropped at save
    d0, i0 = idx.search(q, k)
    p = str(tmp_path / "flat.faissindex")
    idx.save_faiss(p)

    parsed = ff.read_index(p)
    assert parsed["kind"] == "idmap2"
    assert parsed["inner"]["ntotal"] == n - 100
    assert parsed["inner"]["fourcc"] == ("IxF2" if metric == dg.L2
                                         else "IxFI")

    idx2 = dg.Index.load_faiss(p, metric=metric)
    st = idx2.stats()
    assert st["ntotal"] == n - 100 and st["d"] == d
    d1, i1 = idx2.search(q, k)
    np.testing.assert_array_equal(i0, i1)
    np.testing.assert_allclose(d0, d1, rtol=1e-6, atol=1e-6)
    idx.close()
    idx2.close()


@gpu
@pytest.mark.parametrize("metric_name", ["l2", "ip", "cosine"])
def test_ivfflat_roundtrip(tmp_path, metric_name):
    dg = _dg()
    metric = {"l2": dg.L2, "ip": dg.IP, "cosine": dg.COSINE}[metric_name]
    n, d, nq, k, nlist, nprobe = 30000, 96, 64, 10, 64, 16
    base = workload.gen_base(202, n, d)
    q = workload.gen_queries(202, n, d, nq)
    idx = dg.Index(dg.IVF_FLAT, metric, d, nlist=nlist)
    idx.train(base[:8000])
    ids = np.arange(n, dtype=np.int64)
    idx.add(ids, base)
    d0, i0 = idx.search(q, k, nprobe=nprobe)
    p = str(tmp_path / "ivf.faissindex")
    idx.save_faiss(p)

    parsed = ff.read_index(p)
    assert parsed["kind"] == "ivfflat" and parsed["nlist"] == nlist
    assert int(sum(parsed["invlists"]["sizes"])) == n
    # cosine is stored as IP over normalized vectors (flat.cc:88-91)
    assert parsed["header"]["metric"] == (1 if metric == dg.L2 else 0)

    idx2 = dg.Index.load_faiss(p, metric=metric)
    d1, i1 = idx2.search(q, k, nprobe=nprobe)
    np.testing.assert_array_equal(i0, i1)
    np.testing.assert_allclose(d0, d1, rtol=1e-6, atol=1e-6)
    # loaded index keeps the FILE's list assignment: searches with the full
    # sweep must agree too
    d2, i2 = idx.search(q, k, nprobe=nlist)
    d3, i3 = idx2.search(q, k, nprobe=nlist)
    np.testing.assert_array_equal(i2, i3)
    idx.close()
    idx2.close()


@gpu
def test_ivfpq_roundtrip(tmp_path):
    dg = _dg()
    n, d, nq, k, nlist, nprobe, m = 20000, 64, 48, 10, 32, 8, 16
    base = workload.gen_base(303, n, d)
    q = workload.gen_queries(303, n, d, nq)
    idx = dg.Index(dg.IVF_PQ, dg.L2, d, nlist=nlist, m=m)
    idx.train(base[:10000])
    ids = np.arange(n, dtype=np.int64)
    idx.add(ids, base)
    d0, i0 = idx.search(q, k, nprobe=nprobe)
    p = str(tmp_path / "pq.faissindex")
    idx.save_faiss(p)

    parsed = ff.read_index(p)
    assert parsed["kind"] == "ivfpq"
    assert parsed["by_residual"] == 1 and parsed["code_size"] == m
    assert parsed["pq"]["M"] == m and parsed["pq"]["nbits"] == 8
    assert len(parsed["pq"]["centroids"]) == m * 256 * (d // m)

    idx2 = dg.Index.load_faiss(p)
    idx2.m = m
    d1, i1 = idx2.search(q, k, nprobe=nprobe)
    np.testing.assert_array_equal(i0, i1)
    np.testing.assert_allclose(d0, d1, rtol=1e-5, atol=1e-5)
    idx.close()
    idx2.close()


@gpu
def test_golden_generation_and_load(tmp_path):
    """(Re)generate the golden container and compare STRUCTURALLY against
    the committed copy (k-means row grouping uses atomics, so centroid
    last-ulp bits — and, if a boundary row flips early, the converged
    local optimum — are not run-deterministic; the semantic pins are
    structure, the id multiset, and full-sweep search equality).  Always
    leaves a fresh copy under gpurun_out/ for re-committing."""
    dg = _dg()
    n, d, nlist = 2000, 32, 16
    base = workload.gen_base(404, n, d)
    idx = dg.Index(dg.IVF_FLAT, dg.L2, d, nlist=nlist)
    idx.train(base)
    ids = np.arange(n, dtype=np.int64) + 1000
    idx.add(ids, base)
    p = str(tmp_path / "g.faissindex")
    idx.save_faiss(p)
    data = open(p, "rb").read()
    outdir = os.path.join(REPO, "gpurun_out")
    os.makedirs(outdir, exist_ok=True)
    open(os.path.join(outdir, "ivfflat_small.faissindex"), "wb").write(data)
    np.savez(os.path.join(outdir, "ivfflat_small_expect.npz"),
             d=d, nlist=nlist, ids=ids, vectors=base,
             centroids=idx.get_centroids())
    if os.path.exists(GOLDEN):
        # structural + numeric equality with the committed golden.  NOT
        # byte identity: the GPU k-means groups rows with atomics, so the
        # fp summation order (and with it centroid last-ulp bits and the
        # within-list row order) is not run-deterministic.
        a = ff.read_index(GOLDEN)
        b = ff.read_index(p)
        assert a["kind"] == b["kind"] == "ivfflat"
        assert a["header"]["ntotal"] == b["header"]["ntotal"]
        assert a["nlist"] == b["nlist"]
        assert a["quantizer"]["xb"].shape == b["quantizer"]["xb"].shape
        assert sorted(np.concatenate(a["invlists"]["ids"])) == \
            sorted(np.concatenate(b["invlists"]["ids"]))
        # the loaded goldens must SEARCH identically (same engine)
        ga = dg.Index.load_faiss(GOLDEN)
        gb = dg.Index.load_faiss(p)
        q = workload.gen_queries(404, n, d, 16)
        da_, ia_ = ga.search(q, 5, nprobe=nlist)
        db_, ib_ = gb.search(q, 5, nprobe=nlist)
        np.testing.assert_array_equal(ia_, ib_)
        ga.close()
        gb.close()
    idx.close()


def test_parser_rejects_malformed():
    """The independent parser must fail loudly on truncated or corrupt
    containers (and so must dg_load_faiss, via its exception wall — the
    GPU round-trip tests cover that side)."""
    if not os.path.exists(GOLDEN):
        pytest.skip("golden faiss container not generated yet")
    data = open(GOLDEN, "rb").read()
    with pytest.raises(Exception):
        ff.read_index(data[: len(data) // 2])  # truncated mid-invlists
    with pytest.raises(Exception):
        ff.read_index(b"\x00\x01\x02\x03" + data[4:])  # bad fourcc
    with pytest.raises(Exception):
        ff.read_index(data + b"junk")  # trailing bytes
