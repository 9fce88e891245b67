#!/usr/bin/env python3
"""Minimal cohort-prover exerciser for GPU debugging (RNG_COHORT_TRACE=1)."""
import ctypes
import sys
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))
from renegade_amd import load_prover  # noqa: E402
from tests.orc_bindings import OracleLib  # noqa: E402

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)


def main():
    k = int(sys.argv[1]) if len(sys.argv) > 1 else 2
    scale = int(sys.argv[2]) if len(sys.argv) > 2 else 6
    plib = load_prover()
    plib.require_gpu()
    lib = plib.lib
    orc = OracleLib(str(REPO / "oracle" / "liborc.so"))
    lib.rng_testcirc_build.restype = ctypes.c_void_p
    lib.rng_testcirc_build.argtypes = [ctypes.c_uint64] * 2
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]
    lib.rng_preprocess.restype = ctypes.c_void_p
    lib.rng_preprocess.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.rng_prove.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                              ctypes.c_uint64, U64P, U64P]
    lib.rng_prove_cohort.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_uint64, U64P, U64P, U64P, U64P, U64P]
    h = lib.rng_testcirc_build(777, scale)
    n = lib.rng_circ_n(h)
    npub = lib.rng_circ_npub(h)
    print(f"n={n} npub={npub} k={k}", flush=True)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
    lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
    lib.rng_circ_free(h)
    power = max(4, int(n).bit_length())
    ptau = orc.srs_generate_ptau(power, seed=42)
    ctx = plib.init(ptau, (1 << power) + 2)

    class Desc(ctypes.Structure):
        _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                    ("selectors", U64P), ("sigma", U64P),
                    ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

    pk = lib.rng_preprocess(ctx.h, ctypes.byref(Desc(n, npub, ptr(sel), ptr(sigma),
                                                     0, None)))
    assert pk
    wires_all = np.tile(wires, k)
    pubs_all = np.tile(pubs, k)
    seeds = np.arange(100, 100 + k, dtype=np.uint64)
    proofs = np.zeros(157 * k, dtype=np.uint64)
    hints = np.zeros((4 * (int(n) + 2) + 9) * k, dtype=np.uint64)
    rc = lib.rng_prove_cohort(ctx.h, ctypes.c_void_p(pk), k, ptr(wires_all),
                              ptr(pubs_all), ptr(seeds), ptr(proofs), ptr(hints))
    print(f"cohort rc={rc}", flush=True)
    assert rc == 0
    single = np.zeros(157, dtype=np.uint64)
    for p in range(k):
        assert lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(wires), ptr(pubs),
                             ctypes.c_uint64(int(seeds[p])), ptr(single), None) == 0
        assert np.array_equal(single, proofs[157 * p:157 * (p + 1)]), f"proof {p} differs"
    print("cohort == singles, all good", flush=True)


if __name__ == "__main__":
    main()
# (cohort_debug may run under rocprofv3: join pool threads before exit)
import atexit, ctypes as _ct
atexit.register(lambda: _ct.CDLL(str(REPO / "renegade_amd" / "librenegade_prover.so")).rng_shutdown_pool())
