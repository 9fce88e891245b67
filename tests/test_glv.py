"""GLV decomposition (include/bn254_glv.h + k_glv_decompose's math) pinned
against the independent Python derivation (same formulas re-derived from the
curve moduli, as in scripts/gen_glv_params.py)."""
import ctypes

import numpy as np
import pytest

from tests.py_ref import limbs_to_int, int_to_limbs

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

R = 21888242871839275222246405745257275088548364400416034343698204186575808495617
LAM = None
A1 = 9931322734385697763
B1 = -147946756881789319000765030803803410728
A2 = 147946756881789319010696353538189108491
B2 = 9931322734385697763
SH = 320
G1 = (B2 * (1 << SH) + R // 2) // R
G2 = (-B1 * (1 << SH) + R // 2) // R


def py_decompose(k):
    c1 = (G1 * k + (1 << (SH - 1))) >> SH
    c2 = (G2 * k + (1 << (SH - 1))) >> SH
    k1 = k - c1 * A1 - c2 * A2
    k2 = -c1 * B1 - c2 * B2
    return k1, k2


@pytest.fixture(scope="module")
def glv():
    from renegade_amd import load_prover
    lib = load_prover().lib
    lib.rng_glv_decompose.restype = ctypes.c_int
    lib.rng_glv_decompose.argtypes = [U64P] + [U64P] * 4
    return lib


def check(lib, k):
    k4 = np.array(int_to_limbs(k), dtype=np.uint64)
    k1 = np.zeros(4, dtype=np.uint64)
    k2 = np.zeros(4, dtype=np.uint64)
    s1 = np.zeros(1, dtype=np.uint64)
    s2 = np.zeros(1, dtype=np.uint64)
    assert lib.rng_glv_decompose(ptr(k4), ptr(k1), ptr(s1), ptr(k2), ptr(s2)) == 0
    e1, e2 = py_decompose(k)
    got1 = limbs_to_int(k1) * (-1 if s1[0] else 1)
    got2 = limbs_to_int(k2) * (-1 if s2[0] else 1)
    assert got1 == e1 and got2 == e2, f"k={k}"
    # semantic identity: k1 + lambda*k2 == k (mod r); lambda derived from
    # the basis: lambda = -a1 * inverse(b1) mod r (a1 + b1*lambda = 0)
    lam = (-A1 * pow(B1, -1, R)) % R
    assert (got1 + got2 * lam) % R == k % R
    assert abs(got1) < 2**127 and abs(got2) < 2**127


def test_edges(glv):
    lam = (-A1 * pow(B1, -1, R)) % R
    for k in [0, 1, 2, R - 1, R // 2, lam, R - lam, 2**253, 2**128, A2]:
        check(glv, k)


def test_random(glv):
    rng = np.random.default_rng(11)
    for _ in range(300):
        k = int.from_bytes(rng.bytes(32), "little") % R
        check(glv, k)
