"""TurboPlonk oracle prover/verifier round trips on synthetic circuits (CPU).

Mirrors the reference's test strategy: satisfiability at build
time, prove->verify acceptance, tamper rejection.  The circuit tables come
from the product's arithmetization front-end (plonk_circuit.hpp) — the same
tables the GPU prover consumes, so oracle and product provers are fed
identical inputs.
"""
import ctypes

import numpy as np
import pytest

from tests import py_ref as ref

U64P = ctypes.POINTER(ctypes.c_uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


@pytest.fixture(scope="module")
def plib():
    from renegade_amd import load_prover
    return load_prover()  # builder APIs are host-only; no GPU needed


@pytest.fixture(scope="module")
def plonk_setup(plib, orc):
    """Build a mixed circuit + SRS, return everything needed for proving."""
    lib = plib.lib
    lib.rng_testcirc_build.restype = ctypes.c_void_p
    lib.rng_testcirc_build.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]

    h = lib.rng_testcirc_build(1234, 4)
    assert h, "test circuit build failed"
    n = lib.rng_circ_n(h)
    npub = lib.rng_circ_npub(h)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
    lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
    lib.rng_circ_free(h)

    # SRS covering degree n+2
    power = max(4, int(n).bit_length())  # 2^power >= n, degree 2^power+2 >= n+2
    ptau = orc.srs_generate_ptau(power, seed=42)
    max_degree = (1 << power) + 2
    g1, _, _ = orc.srs_parse(ptau, max_degree)
    srs_records = np.ascontiguousarray(g1).reshape(-1)

    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P]
    o.orc_plonk_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]

    pk = o.orc_plonk_preprocess(n, npub, ptr(sel), ptr(sigma), ptr(srs_records),
                                max_degree + 1)
    assert pk, "oracle preprocess failed"
    tau = np.zeros(4, dtype=np.uint64)
    o.orc_derive_tau(42, ptr(tau))
    return dict(n=n, npub=npub, sel=sel, sigma=sigma, wires=wires, pubs=pubs,
                pk=pk, tau=tau, orc=orc, ptau=ptau, max_degree=max_degree)


def prove(s, seed=7):
    proof = np.zeros(157, dtype=np.uint64)
    rc = s["orc"].lib.orc_plonk_prove(ctypes.c_void_p(s["pk"]), ptr(s["wires"]),
                                      ptr(s["pubs"]), ctypes.c_uint64(seed), ptr(proof))
    assert rc == 0, f"prove rc={rc}"
    return proof


def verify(s, proof, pubs=None):
    pubs = s["pubs"] if pubs is None else pubs
    return s["orc"].lib.orc_plonk_verify(ctypes.c_void_p(s["pk"]), ptr(pubs),
                                         ptr(proof), ptr(s["tau"]))


class TestPlonkOracle:
    def test_prove_verify(self, plonk_setup):
        s = plonk_setup
        proof = prove(s)
        assert verify(s, proof) == 1

    def test_deterministic_with_seed(self, plonk_setup):
        s = plonk_setup
        p1, p2 = prove(s, seed=7), prove(s, seed=7)
        assert np.array_equal(p1, p2)
        p3 = prove(s, seed=8)
        assert not np.array_equal(p1, p3)  # blinding differs
        assert verify(s, p3) == 1

    def test_tampered_proof_rejected(self, plonk_setup):
        s = plonk_setup
        base = prove(s)
        # corrupt EVERY proof element: each of the 13 G1 records (x limb)
        # and each of the 10 evaluations (limb 0)
        idxs = [9 * g for g in range(13)] + [117 + 4 * e for e in range(10)]
        for idx in idxs:
            p = base.copy()
            p[idx] ^= np.uint64(1)
            assert verify(s, p) != 1, f"tampered index {idx} accepted"

    def test_wrong_public_inputs_rejected(self, plonk_setup):
        s = plonk_setup
        if s["npub"] == 0:
            pytest.skip("no public inputs")
        proof = prove(s)
        pubs = s["pubs"].copy()
        pubs[0] = (pubs[0] + np.uint64(1))
        assert verify(s, proof, pubs=pubs) != 1

    def test_different_witness_different_proof(self, plib, plonk_setup, orc):
        # circuit with same topology, different seed-> different witness
        lib = plib.lib
        h = lib.rng_testcirc_build(1234, 4)  # same seed/topology as fixture
        n = lib.rng_circ_n(h)
        assert n == plonk_setup["n"]
        lib.rng_circ_free(h)
