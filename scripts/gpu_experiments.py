#!/usr/bin/env python3
"""Ad-hoc GPU tuning experiments (run via gpurun; results inform kernel
design — committed numbers land in DESIGN.md / profiles/).

Legs:
  frmul  — field-mul formulation microbench (CIOS vs 32-bit CIOS vs SOS)
  csweep — 2^20 MSM wall time across window sizes c
  msm    — 2^20 MSM with per-kernel breakdown at the default c
"""
import ctypes
import json
import sys
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from renegade_amd import load_prover  # noqa: E402
from tests.orc_bindings import OracleLib  # noqa: E402

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)


def main():
    legs = sys.argv[1:] or ["frmul", "csweep", "msm"]
    plib = load_prover()
    plib.require_gpu()
    lib = plib.lib
    out = {}

    if "frmul" in legs:
        lib.rng_bench_frmul.restype = ctypes.c_double
        lib.rng_bench_frmul.argtypes = [ctypes.c_int, ctypes.c_uint32,
                                        ctypes.c_uint32, ctypes.c_int, ctypes.c_int]
        # 2048 blocks x 256 threads x 16384 muls each
        blocks, iters = 2048, 16384
        names = {0: "cios_u64", 1: "cios_u32", 2: "sos_u64"}
        res = {}
        for variant in (0, 1, 2):
            for dep in (0, 1):
                ms = lib.rng_bench_frmul(variant, blocks, iters, dep, 3)
                gmuls = blocks * 256 * iters / (ms / 1e3) / 1e9
                res[f"{names[variant]}_{'dep' if dep else 'ilp4'}"] = {
                    "ms": round(ms, 3), "gmul_per_s": round(gmuls, 1)}
        out["frmul"] = res
        print(json.dumps({"frmul": res}), flush=True)

    orc = OracleLib(str(REPO / "oracle" / "liborc.so"))
    power = 12
    ptau = orc.srs_generate_ptau(power, seed=42)
    g1, _, _ = orc.srs_parse(ptau, (1 << power) + 2)
    ctx = plib.init(ptau, (1 << power) + 2)
    n = 1 << 20
    npts = g1.shape[0]
    reps = (n + npts - 1) // npts
    bases8 = np.ascontiguousarray(
        np.tile(np.ascontiguousarray(g1[:, :8]), (reps, 1))[:n].reshape(-1))
    rng = np.random.default_rng(12345)
    scalars = rng.integers(0, 1 << 64, size=4 * n, dtype=np.uint64)
    scalars[3::4] &= (1 << 61) - 1
    db = ctx.dbuf_from(bases8)
    ds = ctx.dbuf_from(scalars)

    ref = None
    if "csweep" in legs:
        res = {}
        for c in (11, 12, 13, 14, 15, 16):
            r = ctx.msm_dev(db, ds, n, window_c=c)
            if ref is None:
                ref = r
            assert np.array_equal(r, ref), f"c={c} result mismatch"
            ctx.sync()
            t0 = time.perf_counter()
            for _ in range(5):
                ctx.msm_dev(db, ds, n, window_c=c)
            ms = (time.perf_counter() - t0) / 5 * 1e3
            res[f"c{c}"] = {"ms": round(ms, 3),
                            "breakdown": {k: round(v, 3)
                                          for k, v in plib.msm_last_times().items()}}
        out["csweep"] = res
        print(json.dumps({"csweep": res}), flush=True)

    if "msm" in legs:
        for _ in range(2):
            ctx.msm_dev(db, ds, n, window_c=16)
        kt = []
        t0 = time.perf_counter()
        for _ in range(5):
            ctx.msm_dev(db, ds, n, window_c=16)
            kt.append(plib.msm_last_times())
        ms = (time.perf_counter() - t0) / 5 * 1e3
        out["msm"] = {"ms": round(ms, 3),
                      "breakdown": {k: round(float(np.mean([t[k] for t in kt])), 3)
                                    for k in kt[0]}}
        print(json.dumps({"msm": out["msm"]}), flush=True)

    Path("gpurun_out").mkdir(exist_ok=True)
    with open("gpurun_out/experiments.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()

import atexit as _atexit
import ctypes as _ct
_atexit.register(lambda: _ct.CDLL(str(REPO / "renegade_amd" / "librenegade_prover.so")).rng_shutdown_pool())
