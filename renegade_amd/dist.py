"""Multi-GPU sharding helpers.

The hot path shards two ways:
 - proof jobs are embarrassingly parallel (the headline metric): each rank
   proves independently — no data-path collective (bench.py default mode);
 - a single large MSM base-splits by index: each rank runs full Pippenger on
   its shard of the base/scalar arrays, then ONE exchange of the per-rank
   partial sums (G1 points; EC add is not an RCCL reduce op, so this is an
   all_gather of 72-byte records + a local EC fold — latency-bound, a few
   hundred bytes over xGMI).

The exchange/combine logic is torch.distributed-backend-agnostic, so the
world_size-2 CPU tests run it over gloo with the oracle MSM as the compute
leg, and bench.py --mode msm-shard runs it over RCCL with the HIP MSM.
"""
import numpy as np


def shard_bounds(n, world, rank):
    """Contiguous index shard [lo, hi) for this rank."""
    per = (n + world - 1) // world
    lo = min(rank * per, n)
    hi = min(lo + per, n)
    return lo, hi


def combine_shard_results(dist_mod, record9: np.ndarray, add_fn):
    """All-gather per-rank 9-u64 affine MSM partials and EC-fold them.

    add_fn(a9, b9) -> 9-u64 record (product host EC add, or the oracle's in
    CPU tests).  Every rank returns the combined record (the gather is
    symmetric).
    """
    import torch
    t = torch.from_numpy(record9.view(np.int64).copy())
    on_gpu = dist_mod.get_backend() == "nccl"
    if on_gpu:
        t = t.cuda()  # NCCL collectives need device tensors
    out = [torch.zeros_like(t) for _ in range(dist_mod.get_world_size())]
    dist_mod.all_gather(out, t)
    if on_gpu:
        out = [r.cpu() for r in out]
    acc = out[0].numpy().view(np.uint64).copy()
    for r in out[1:]:
        acc = add_fn(acc, r.numpy().view(np.uint64).copy())
    return acc
