"""A/B the PQ scan default vs COOP + new select width on one cfg D build."""
import argparse, os, sys, time
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path[:0] = [REPO, os.path.join(REPO, "dingo-store_amd")]
import torch
import bench as B

args = argparse.Namespace(n=100_000_000, d=768, nlist=16384, nprobe=64, k=10,
                          batch=4096, seed=4244, kind="ivf_pq", m=96)
dev = torch.device("cuda:0")
idx, _ = B.build_index(args, 0, 1, dev)
q = B.gen_queries_device(args.seed, args.n, args.d, args.batch, dev)
dist = torch.empty((args.batch, args.k), dtype=torch.float32, device=dev)
ids = torch.empty((args.batch, args.k), dtype=torch.int64, device=dev)
ref = None
for rpv in ("4", "17"):
    os.environ["DG_PQ_RPV"] = rpv
    for _ in range(2):
        B.merged_step(idx, q, args.k, args.nprobe, dist, ids, 1, None)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(5):
        B.merged_step(idx, q, args.k, args.nprobe, dist, ids, 1, None)
    torch.cuda.synchronize()
    dt = time.time() - t0
    st = idx.stats()
    same = None
    if ref is None:
        ref = ids.clone()
    else:
        same = bool(torch.equal(ref, ids))
    print(f"RPV {rpv}: ms/step {dt/5*1000:.2f} QPS {args.batch*5/dt:.0f} "
          f"scan_ms {st['last_scan_ms']:.2f} coarse_ms {st['last_coarse_ms']:.2f} "
          f"ids_match {same}", flush=True)
idx.close()
