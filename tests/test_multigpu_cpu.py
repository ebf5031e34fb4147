"""Multi-process coverage of the distributed search path on CPU (gloo,
world_size=2): the bench's row-sharded search + all-gather + merge must
reproduce the single-index result exactly.  Rank-local search uses the
oracle (tests may); the merge logic under test is the same code bench.py
runs over RCCL on the GPU node (bench.torch_merge / dingostore.merge_topk).
"""
import multiprocessing as mp
import os
import sys

import numpy as np
import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
sys.path[:0] = [REPO, os.path.join(REPO, "oracle"),
                os.path.join(REPO, "dingo-store_amd")]


def _worker(rank, world, port, ret):
    import torch
    import torch.distributed as dist
    sys.path[:0] = [REPO, os.path.join(REPO, "oracle"),
                    os.path.join(REPO, "dingo-store_amd")]
    import pyoracle as orc
    import workload
    from bench import torch_merge

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)

    n, d, nlist, nprobe, k, nq = 8000, 32, 32, 8, 10, 64
    seed = 555
    # shared centroids: trained deterministically on the first rows
    train = workload.gen_base(seed, n, d, 0, n // 2)
    cents = orc.kmeans(orc.L2, train, nlist)
    # rank's row shard
    r0, r1 = rank * n // world, (rank + 1) * n // world
    shard = workload.gen_base(seed, n, d, r0, r1)
    ids = np.arange(r0, r1, dtype=np.int64)
    assign = orc.ivf_assign(orc.L2, shard, cents)
    off, gv, gi = orc.ivf_build(shard, ids, nlist, assign)
    q = workload.gen_queries(seed, n, d, nq)
    ld, li = orc.ivf_search(orc.L2, cents, off, gv, gi, q, k, nprobe)

    # all-gather per-shard top-k (the xGMI exchange, gloo here)
    dt = torch.from_numpy(ld)
    it = torch.from_numpy(li)
    gd = [torch.empty_like(dt) for _ in range(world)]
    gi_ = [torch.empty_like(it) for _ in range(world)]
    dist.all_gather(gd, dt)
    dist.all_gather(gi_, it)
    md, mi = torch_merge(gd, gi_, k)

    if rank == 0:
        # reference: single whole-database search
        base = workload.gen_base(seed, n, d)
        a_all = orc.ivf_assign(orc.L2, base, cents)
        o2, v2, i2 = orc.ivf_build(base, None, nlist, a_all)
        wd, wi = orc.ivf_search(orc.L2, cents, o2, v2, i2, q, k, nprobe)
        ret["ids_equal"] = bool(np.array_equal(mi.numpy(), wi))
        ret["dist_close"] = bool(
            np.allclose(md.numpy(), wd, rtol=1e-5, atol=1e-5))
        # numpy merge helper agrees with the torch merge
        import dingostore as dgs
        nd_, ni_ = dgs.merge_topk(
            np.stack([t.numpy() for t in gd]),
            np.stack([t.numpy() for t in gi_]), k)
        ret["numpy_merge_equal"] = bool(np.array_equal(ni_, mi.numpy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("world,port", [(2, 29781), (3, 29782)])
def test_row_shard_merge_gloo(world, port):
    # world=3 exercises uneven row shards (8000 % 3 != 0)
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    ret = mgr.dict()
    ps = [ctx.Process(target=_worker, args=(r, world, port, ret))
          for r in range(world)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(timeout=180)
        assert p.exitcode == 0
    assert ret["ids_equal"]
    assert ret["dist_close"]
    assert ret["numpy_merge_equal"]
