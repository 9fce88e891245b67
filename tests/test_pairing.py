"""BN254 pairing (product host path behind rng_verify): bilinearity pins on
oracle-generated points (CPU).  e(aG, H) == e(G, aH); e(aG, bH) == e(abG, H);
mismatches rejected."""
import ctypes
import random

import numpy as np
import pytest

from tests import py_ref as ref

U64P = ctypes.POINTER(ctypes.c_uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


@pytest.fixture(scope="module")
def pl(orc):
    from renegade_amd import load_prover
    plib = load_prover()
    plib.lib.rng_pairing_check.argtypes = [U64P, U64P, U64P, U64P]
    plib.lib.rng_pairing_check.restype = ctypes.c_int
    orc.lib.orc_g2_mul_gen.argtypes = [U64P, U64P]
    return plib


def g1_mul(orc, k):
    return orc.g1_mul(orc.g1_generator(), k)


def g2_mul(orc, k):
    s = np.zeros(4, dtype=np.uint64)
    s[:] = ref.int_to_limbs(k)
    out = np.zeros(16, dtype=np.uint64)
    orc.lib.orc_g2_mul_gen(ptr(s), ptr(out))
    return out


class TestPairing:
    def test_bilinearity(self, pl, orc):
        rng = random.Random(31)
        a = rng.randrange(1, ref.R)
        b = rng.randrange(1, ref.R)
        G = g1_mul(orc, 1)
        H = g2_mul(orc, 1)
        aG, bG, abG = g1_mul(orc, a), g1_mul(orc, b), g1_mul(orc, a * b % ref.R)
        aH, bH = g2_mul(orc, a), g2_mul(orc, b)
        chk = pl.lib.rng_pairing_check
        assert chk(ptr(aG), ptr(H), ptr(G), ptr(aH)) == 1      # e(aG,H)=e(G,aH)
        assert chk(ptr(aG), ptr(bH), ptr(abG), ptr(H)) == 1    # e(aG,bH)=e(abG,H)
        assert chk(ptr(bG), ptr(aH), ptr(abG), ptr(H)) == 1
        assert chk(ptr(aG), ptr(H), ptr(G), ptr(bH)) == 0      # a != b
        assert chk(ptr(aG), ptr(bH), ptr(G), ptr(H)) == 0

    def test_srs_ratio(self, pl, orc):
        # mirror of the reference's SRS pairing-ratio test (srs.rs:236-266):
        # e(tau^i G, tau H) == e(tau^(i+1) G, H)
        ptau = orc.srs_generate_ptau(6, seed=42)
        g1, h, bh = orc.srs_parse(ptau, (1 << 6) + 2)
        chk = pl.lib.rng_pairing_check
        for i in [0, 3, 40]:
            a = np.ascontiguousarray(g1[i])
            b = np.ascontiguousarray(g1[i + 1])
            assert chk(ptr(a), ptr(bh), ptr(b), ptr(h)) == 1
        a = np.ascontiguousarray(g1[0])
        b = np.ascontiguousarray(g1[2])
        assert chk(ptr(a), ptr(bh), ptr(b), ptr(h)) == 0
