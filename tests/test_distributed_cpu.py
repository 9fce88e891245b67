"""World-size-2 gloo tests of the multi-GPU shard/exchange logic on CPU
(no GPU — the compute leg is the oracle MSM, the exchange is
the same torch.distributed code bench.py uses over RCCL)."""
import ctypes
import os
import random

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


def _worker(rank, world, port, n, seed, result_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    from pathlib import Path
    repo = Path(__file__).resolve().parent.parent
    sys.path.insert(0, str(repo))
    from tests.orc_bindings import OracleLib
    from tests import py_ref as ref
    from renegade_amd.dist import shard_bounds, combine_shard_results

    orc = OracleLib(str(repo / "oracle" / "liborc.so"))

    # same synthetic inputs on every rank (seeded)
    power = 8
    ptau = orc.srs_generate_ptau(power, seed=42)
    g1, _, _ = orc.srs_parse(ptau, (1 << power) + 2)
    rng = random.Random(seed)
    bases9 = np.ascontiguousarray(np.tile(g1, ((n + g1.shape[0] - 1) // g1.shape[0], 1))[:n]).reshape(-1)
    scalars = np.zeros(4 * n, dtype=np.uint64)
    for i in range(n):
        scalars[4 * i:4 * i + 4] = ref.int_to_limbs(rng.randrange(ref.R))

    lo, hi = shard_bounds(n, world, rank)
    part = orc.msm(np.ascontiguousarray(bases9.reshape(n, 9)[lo:hi]).reshape(-1),
                   np.ascontiguousarray(scalars.reshape(n, 4)[lo:hi]).reshape(-1),
                   hi - lo)

    def add_fn(a, b):
        out = np.zeros(9, dtype=np.uint64)
        orc.lib.orc_g1_add(ptr(a), ptr(b), ptr(out))
        return out

    combined = combine_shard_results(dist, part, add_fn)
    if rank == 0:
        full = orc.msm(bases9, scalars, n)
        result_q.put((combined.tolist(), full.tolist()))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_sharded_msm_exchange_world2():
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29000 + random.randrange(500)
    n = 300
    procs = [ctx.Process(target=_worker, args=(r, 2, port, n, 1234, q)) for r in range(2)]
    for p in procs:
        p.start()
    combined, full = q.get(timeout=240)
    for p in procs:
        p.join(timeout=60)
    assert combined == full, "sharded MSM combine != full MSM"


def test_shard_bounds_cover():
    from renegade_amd.dist import shard_bounds
    for n in [1, 7, 300, 1 << 20]:
        for world in [1, 2, 8]:
            spans = [shard_bounds(n, world, r) for r in range(world)]
            assert spans[0][0] == 0 and spans[-1][1] == n
            for (a, b), (c, d) in zip(spans, spans[1:]):
                assert b == c
