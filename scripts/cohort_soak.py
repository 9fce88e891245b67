#!/usr/bin/env python3
"""Ad-hoc cohort parity soak: many seeds x cohort sizes vs the single-proof
path on the headline circuit (run via gpurun; results recorded in DESIGN)."""
import ctypes
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
    plib = load_prover()
    plib.require_gpu()
    lib = plib.lib
    orc = OracleLib(str(REPO / "oracle" / "liborc.so"))
    lib.rng_circ_build_settlement.restype = ctypes.c_void_p
    lib.rng_circ_build_settlement.argtypes = [ctypes.c_uint64]
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
    h = lib.rng_circ_build_settlement(42)
    n = lib.rng_circ_n(h)
    npub = lib.rng_circ_npub(h)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(npub * 4, dtype=np.uint64)
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
    rng = np.random.default_rng(2024)
    t0 = time.time()
    checked = 0
    single = np.zeros(157, dtype=np.uint64)
    for trial in range(200):
        k = int(rng.choice([2, 3, 5, 8, 13, 16, 32, 48, 64]))
        seeds = rng.integers(0, 1 << 62, size=k, dtype=np.uint64)
        proofs = np.zeros(157 * k, dtype=np.uint64)
        rc = lib.rng_prove_cohort(ctx.h, ctypes.c_void_p(pk), k,
                                  ptr(np.tile(wires, k)), ptr(np.tile(pubs, k)),
                                  ptr(seeds), ptr(proofs), None)
        assert rc == 0, f"trial {trial} k={k} rc={rc}"
        # spot-check 2 random positions per cohort against the single path
        for p in rng.choice(k, size=min(2, k), replace=False):
            assert lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(wires), ptr(pubs),
                                 ctypes.c_uint64(int(seeds[p])), ptr(single),
                                 None) == 0
            assert np.array_equal(single, proofs[157 * p:157 * (p + 1)]), \
                f"trial {trial} k={k} pos {p} MISMATCH"
            checked += 1
        if time.time() - t0 > 240:
            break
    print(f"soak ok: {trial + 1} cohorts (k in 2..64), {checked} positions "
          f"bit-exact vs single in {time.time() - t0:.0f}s", flush=True)


if __name__ == "__main__":
    main()
