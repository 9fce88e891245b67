#!/usr/bin/env python3
"""Generate golden proof fixtures under tests/golden/.

The fixtures pin the ENTIRE protocol byte-for-byte across rounds (circuit
construction, arithmetization, SRS derivation, transcript, blinder DRBG,
prover algebra): any unintentional change to any layer shows up as a fixture
mismatch.  Generated from the CPU oracle (no GPU needed); the GPU prover is
bit-exact vs the oracle by the parity tests.
"""
import ctypes
import sys
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))
U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)


def main():
    from renegade_amd import load_prover
    from tests.orc_bindings import OracleLib
    plib = load_prover()
    lib = plib.lib
    orc = OracleLib(str(REPO / "oracle" / "liborc.so"))
    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P]
    out_dir = REPO / "tests" / "golden"
    out_dir.mkdir(exist_ok=True)

    # (name, builder, extra builder args); covers every gadget family:
    # rotation+partial commitments, links, Schnorr, ElGamal, notes
    targets = [("settlement", "rng_circ_build_settlement", ()),
               ("vbc", "rng_circ_build_vbc", ()),
               ("validity", "rng_circ_build_validity", (0,)),
               ("ff_validity", "rng_circ_build_ff_validity", (0,)),
               ("nob_validity", "rng_circ_build_nob_validity", ()),
               ("fee_private_protocol", "rng_circ_build_fee_private_protocol", ()),
               ("io_validity", "rng_circ_build_io_validity", ())]
    for name, builder, extra in targets:
        fn = getattr(lib, builder)
        fn.restype = ctypes.c_void_p
        fn.argtypes = [ctypes.c_uint64] * (1 + len(extra))
        lib.rng_circ_n.restype = ctypes.c_uint64
        lib.rng_circ_n.argtypes = [ctypes.c_void_p]
        lib.rng_circ_npub.restype = ctypes.c_uint64
        lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
        lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
        lib.rng_circ_free.argtypes = [ctypes.c_void_p]
        h = fn(42, *extra)
        assert h
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
        max_degree = (1 << power) + 2
        g1, _, _ = orc.srs_parse(ptau, max_degree)
        srs_records = np.ascontiguousarray(g1).reshape(-1)
        pk = o.orc_plonk_preprocess(n, npub, ptr(sel), ptr(sigma), ptr(srs_records),
                                    max_degree + 1)
        assert pk
        proof = np.zeros(157, dtype=np.uint64)
        assert o.orc_plonk_prove(ctypes.c_void_p(pk), ptr(wires), ptr(pubs),
                                 ctypes.c_uint64(7), ptr(proof)) == 0
        np.save(out_dir / f"{name}_proof_cseed42_bseed7.npy", proof)
        np.save(out_dir / f"{name}_pubs_cseed42.npy", pubs)
        print(f"{name}: n={n} npub={npub} proof[0]={proof[0]:#x}")


if __name__ == "__main__":
    main()
