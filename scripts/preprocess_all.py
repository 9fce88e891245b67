#!/usr/bin/env python3
"""Preprocess ALL 20 circuits on the GPU and time it (the
reference's startup preprocesses every circuit's proving key on the CPU —
a minutes-scale warmup; here it is seconds).  Run on a GPU box:

    python scripts/preprocess_all.py
"""
import ctypes
import sys
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))
U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

BUILDERS = [
    ("valid_balance_create", "rng_circ_build_vbc", ()),
    ("valid_deposit", "rng_circ_build_valid_deposit", ()),
    ("valid_withdrawal", "rng_circ_build_valid_withdrawal", ()),
    ("valid_order_cancellation", "rng_circ_build_valid_order_cancellation", ()),
    ("intent_and_balance_validity", "rng_circ_build_validity", (0,)),
    ("intent_and_balance_first_fill", "rng_circ_build_ff_validity", (0,)),
    ("intent_only_validity", "rng_circ_build_io_validity", ()),
    ("intent_only_first_fill", "rng_circ_build_ioff", ()),
    ("new_output_balance", "rng_circ_build_nob_validity", ()),
    ("output_balance_validity", "rng_circ_build_ob_validity", (0,)),
    ("ib_private_settlement", "rng_circ_build_settlement", ()),
    ("ib_public_settlement", "rng_circ_build_public_settlement", ()),
    ("ib_bounded_settlement", "rng_circ_build_ib_bounded_settlement", ()),
    ("io_public_settlement", "rng_circ_build_io_settlement", ()),
    ("io_bounded_settlement", "rng_circ_build_io_bounded_settlement", ()),
    ("valid_note_redemption", "rng_circ_build_note_redemption", ()),
    ("fee_public_relayer", "rng_circ_build_fee_public_relayer", ()),
    ("fee_public_protocol", "rng_circ_build_fee_public_protocol", ()),
    ("fee_private_relayer", "rng_circ_build_fee_private_relayer", ()),
    ("fee_private_protocol", "rng_circ_build_fee_private_protocol", ()),
]


class Desc(ctypes.Structure):
    _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                ("selectors", U64P), ("sigma", U64P),
                ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]


def main():
    from renegade_amd import load_prover
    from tests.orc_bindings import OracleLib
    plib = load_prover()
    plib.require_gpu()
    lib = plib.lib
    orc = OracleLib(str(REPO / "oracle" / "liborc.so"))
    t_srs = time.perf_counter()
    ptau = orc.srs_generate_ptau(15, seed=42)
    max_degree = (1 << 15) + 2
    ctx = plib.init(ptau, max_degree)
    print(f"SRS generate+parse+upload (2^15): {time.perf_counter() - t_srs:.2f} s")
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]
    lib.rng_preprocess.restype = ctypes.c_void_p
    lib.rng_preprocess.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.rng_pk_free.argtypes = [ctypes.c_void_p]

    total_build = total_pp = 0.0
    for name, builder, extra in BUILDERS:
        fn = getattr(lib, builder)
        fn.restype = ctypes.c_void_p
        fn.argtypes = [ctypes.c_uint64] * (1 + len(extra))
        t0 = time.perf_counter()
        h = fn(42, *extra)
        assert h, name
        n = lib.rng_circ_n(h)
        npub = lib.rng_circ_npub(h)
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sigma = np.zeros(5 * n, dtype=np.uint64)
        wires = np.zeros(5 * n * 4, dtype=np.uint64)
        pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
        lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
        lib.rng_circ_free(h)
        t1 = time.perf_counter()
        desc = Desc(n, npub, ptr(sel), ptr(sigma), 0, None)
        pk = lib.rng_preprocess(ctx.h, ctypes.byref(desc))
        assert pk, name
        t2 = time.perf_counter()
        lib.rng_pk_free(pk)
        total_build += t1 - t0
        total_pp += t2 - t1
        print(f"{name:32s} n={n:6d} build {1e3*(t1-t0):7.1f} ms  "
              f"preprocess {1e3*(t2-t1):7.1f} ms")
    print(f"\nTOTAL: circuit build (host) {total_build:.2f} s + "
          f"GPU preprocess {total_pp:.2f} s for all 20 circuits")


if __name__ == "__main__":
    main()
