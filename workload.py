"""workload.py — the synthetic-data protocol shared by tests, bench.py and
the CPU baseline (restates BASELINE.md's data plan).

Base vectors: fp32 i.i.d. uniform[0,1), the same distribution the
reference's own tests generate (test_vector_index_recall_flat.cc:110-119).
Queries: perturbed database vectors (base[i % n] + N(0, 0.05)) so recall@10
is non-trivial; ground truth by exact scan on identical data.

Generation is chunked and seeded per chunk (numpy Philox via SeedSequence
spawn keys) so a multi-GPU rank can generate exactly its own row range
without materializing the rest.  NOTHING here reads /root/reference.
"""
import numpy as np

CHUNK = 1 << 18  # 262144 rows per chunk


def gen_base(seed, n, d, row_start=0, row_end=None, out=None):
    """Rows [row_start, row_end) of the n x d base set for `seed`."""
    if row_end is None:
        row_end = n
    if out is None:
        out = np.empty((row_end - row_start, d), np.float32)
    c0, c1 = row_start // CHUNK, (row_end + CHUNK - 1) // CHUNK
    for c in range(c0, c1):
        lo, hi = max(c * CHUNK, row_start), min((c + 1) * CHUNK, row_end, n)
        if hi <= lo:
            continue
        rng = np.random.Generator(np.random.Philox(key=[(seed << 32) | 0xBA5E, c]))
        chunk = rng.random((min(CHUNK, n - c * CHUNK), d), dtype=np.float32)
        out[lo - row_start:hi - row_start] = chunk[lo - c * CHUNK:hi - c * CHUNK]
    return out


def gen_queries(seed, n, d, nq):
    """nq queries = base rows (cycled) + N(0, 0.05) noise."""
    idx = np.arange(nq, dtype=np.int64) % n
    q = np.empty((nq, d), np.float32)
    # gather the needed base rows chunk by chunk
    order = np.argsort(idx, kind="stable")
    sorted_idx = idx[order]
    pos = 0
    while pos < nq:
        c = sorted_idx[pos] // CHUNK
        end = pos
        while end < nq and sorted_idx[end] // CHUNK == c:
            end += 1
        rng = np.random.Generator(np.random.Philox(key=[(seed << 32) | 0xBA5E, int(c)]))
        chunk = rng.random((min(CHUNK, n - c * CHUNK), d), dtype=np.float32)
        q[order[pos:end]] = chunk[sorted_idx[pos:end] - c * CHUNK]
        pos = end
    nrng = np.random.Generator(np.random.Philox(key=[(seed << 32) | 0x90153, 0]))
    q += nrng.normal(0.0, 0.05, size=q.shape).astype(np.float32)
    return q


def train_sample(seed, n, d, n_train):
    """Deterministic training subsample: the first n_train base rows.

    (faiss subsamples internally with its own seeded perm — oracle.c
    restates that; this is just which rows are OFFERED to Train.)
    """
    return gen_base(seed, n, d, 0, min(n_train, n))


# BASELINE.json configs as concrete parameter sets
CONFIGS = {
    "A": dict(kind="flat", metric=0, n=100_000, d=128, nq=1, k=10, seed=4242),
    "B": dict(kind="flat", metric=0, n=1_000_000, d=768, nq=256, k=10,
              seed=4243),
    "C": dict(kind="ivf_flat", metric=0, n=10_000_000, d=768, nlist=4096,
              nprobe=32, nq=1024, k=10, seed=4244),
    "D": dict(kind="ivf_pq", metric=0, n=100_000_000, d=768, m=96, nbits=8,
              nlist=16384, nprobe=64, nq=4096, k=10, seed=4245),
    "E": dict(kind="ivf_flat", metric=0, n=10_000_000, d=768, nlist=4096,
              nprobe=32, nq=8192, k=10, seed=4244),
}
