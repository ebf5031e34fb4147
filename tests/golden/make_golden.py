"""Generate committed golden fixtures from the oracle (small sizes; the
oracle itself is pinned against oracle/_ref + numpy f64 + reference
behavioral properties in tests/test_oracle.py).  Run from repo root:
    python tests/golden/make_golden.py
"""
import os
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))
sys.path[:0] = [REPO, os.path.join(REPO, "oracle")]

import pyoracle as orc  # noqa: E402
import workload  # noqa: E402

SEED, N, D, NQ, K, NLIST, NPROBE = 31337, 5000, 64, 32, 10, 32, 8

base = workload.gen_base(SEED, N, D)
q = workload.gen_queries(SEED, N, D, NQ)
fd, fi = orc.flat_search(orc.L2, base, q, K)
cents = orc.kmeans(orc.L2, base, NLIST)
assign = orc.ivf_assign(orc.L2, base, cents)
off, gv, gi = orc.ivf_build(base, None, NLIST, assign)
vd, vi = orc.ivf_search(orc.L2, cents, off, gv, gi, q, K, NPROBE)

# IVF-PQ golden (m=8 over d=64)
M = 8
residuals = base - cents[assign]
cb = orc.pq_train(residuals, M)
codes = orc.ivfpq_encode(base, assign, cents, cb)
gcodes = np.empty_like(codes)
cursor = off[:-1].copy()
for i in range(N):
    gcodes[cursor[assign[i]]] = codes[i]
    cursor[assign[i]] += 1
pd, pi = orc.ivfpq_search(orc.L2, cents, off, gcodes, gi, cb, q, K, NPROBE)

# range-search golden at the median 5th-NN distance
radius = float(np.median(fd[:, 5]))
rl, rd, ri = orc.flat_range_search(orc.L2, base, q, radius)

np.savez_compressed(
    os.path.join(HERE, "oracle_golden.npz"),
    seed=SEED, n=N, d=D, nq=NQ, k=K, nlist=NLIST, nprobe=NPROBE,
    flat_dist=fd, flat_ids=fi, centroids=cents, ivf_dist=vd, ivf_ids=vi,
    pq_m=M, codebooks=cb, pq_dist=pd, pq_ids=pi,
    range_radius=radius, range_lims=rl, range_dist=rd, range_ids=ri)
print("wrote oracle_golden.npz",
      os.path.getsize(os.path.join(HERE, "oracle_golden.npz")), "bytes")
