"""Host-side BN254 G1 affine addition export pinned against the independent
Python model (the same op folds MSM partials on the host)."""
import ctypes

import numpy as np

from tests import py_ref as ref

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)


def rec(point):
    out = np.zeros(9, dtype=np.uint64)
    if point is None:
        out[8] = 1
        return out
    out[:4] = ref.int_to_limbs(ref.to_mont(point[0], ref.Q))
    out[4:8] = ref.int_to_limbs(ref.to_mont(point[1], ref.Q))
    return out


def unrec(r):
    if r[8]:
        return None
    return (ref.from_mont(ref.limbs_to_int(r[:4]), ref.Q),
            ref.from_mont(ref.limbs_to_int(r[4:8]), ref.Q))


def test_g1_add_affine_vs_python():
    from renegade_amd import load_prover
    lib = load_prover().lib
    lib.rng_g1_add_affine.argtypes = [U64P, U64P, U64P]
    G = (1, 2)
    pts = [None, G, ref.ec_mul(G, 5), ref.ec_mul(G, 7),
           ref.ec_mul(G, ref.R - 1)]  # includes -G (a + (-a) = identity)
    for a in pts:
        for b in pts:
            out = np.zeros(9, dtype=np.uint64)
            lib.rng_g1_add_affine(ptr(rec(a)), ptr(rec(b)), ptr(out))
            assert unrec(out) == ref.ec_add(a, b), (a, b)
