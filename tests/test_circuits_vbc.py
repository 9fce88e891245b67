"""`Valid Balance Create` circuit (BASELINE config #1) + Poseidon2 parity (CPU).

Pins:
 - Poseidon2: product native == oracle restatement == pure-Python model built
   from the in-repo constants (crypto/src/hash/constants.rs ->
   include/poseidon2_constants.h).
 - VBC circuit: satisfiable on the fixed-seed witness/statement
   (valid_balance_create.rs test_helpers), oracle prove -> verify round trip,
   wrong-publics rejection.
"""
import ctypes
import re
from pathlib import Path

import numpy as np
import pytest

from tests import py_ref as ref

U64P = ctypes.POINTER(ctypes.c_uint64)
REPO = Path(__file__).resolve().parent.parent


def ptr(a):
    return a.ctypes.data_as(U64P)


# ---- pure-python Poseidon2 (third, independent implementation) ----

def _load_constants():
    src = (REPO / "include/poseidon2_constants.h").read_text()
    nums = re.findall(r"\{ (0x[0-9a-f]+)ULL, (0x[0-9a-f]+)ULL, (0x[0-9a-f]+)ULL, (0x[0-9a-f]+)ULL \}", src)
    vals = [sum(int(x, 16) << (64 * i) for i, x in enumerate(q)) for q in nums]
    assert len(vals) == 80
    full = [vals[3 * r:3 * r + 3] for r in range(8)]
    partial = vals[24:]
    return full, partial


FULL_RC, PARTIAL_RC = _load_constants()
R = ref.R


def py_permute(st):
    def ext(s):
        t = sum(s) % R
        return [(x + t) % R for x in s]

    def internal(s):
        t = sum(s) % R
        s = [s[0], s[1], 2 * s[2] % R]
        return [(x + t) % R for x in s]

    p5 = lambda x: pow(x, 5, R)
    st = ext(st)
    for r in range(4):
        st = [(x + c) % R for x, c in zip(st, FULL_RC[r])]
        st = ext([p5(x) for x in st])
    for r in range(56):
        st[0] = p5((st[0] + PARTIAL_RC[r]) % R)
        st = internal(st)
    for r in range(4, 8):
        st = [(x + c) % R for x, c in zip(st, FULL_RC[r])]
        st = ext([p5(x) for x in st])
    return st


def py_poseidon_hash(inputs):
    st = [0, 0, 0]
    idx = 0
    for x in inputs:
        if idx == 2:
            st = py_permute(st)
            idx = 0
        st[1 + idx] = (st[1 + idx] + x) % R
        idx += 1
    st = py_permute(st)
    return st[1]


@pytest.fixture(scope="module")
def plib():
    from renegade_amd import load_prover
    return load_prover()


class TestPoseidon2:
    @pytest.mark.parametrize("nin", [1, 2, 3, 5, 12, 13])
    def test_three_way_parity(self, plib, orc, nin):
        import random
        rng = random.Random(50 + nin)
        vals = [rng.randrange(ref.R) for _ in range(nin)]
        mont = np.zeros(4 * nin, dtype=np.uint64)
        for i, v in enumerate(vals):
            mont[4 * i:4 * i + 4] = ref.int_to_limbs(ref.to_mont(v, ref.R))
        # product
        out_p = np.zeros(4, dtype=np.uint64)
        plib.lib.rng_poseidon_hash(ptr(mont), ctypes.c_uint64(nin), ptr(out_p))
        # oracle
        out_o = np.zeros(4, dtype=np.uint64)
        orc.lib.orc_poseidon2_hash(ptr(mont), ctypes.c_uint64(nin), ptr(out_o))
        assert np.array_equal(out_p, out_o)
        # python
        expect = py_poseidon_hash(vals)
        got = ref.from_mont(ref.limbs_to_int(out_p), ref.R)
        assert got == expect


@pytest.fixture(scope="module")
def vbc(plib, orc):
    lib = plib.lib
    lib.rng_circ_build_vbc.restype = ctypes.c_void_p
    lib.rng_circ_build_vbc.argtypes = [ctypes.c_uint64]
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]
    h = lib.rng_circ_build_vbc(42)
    assert h, "VBC circuit build failed (unsatisfied?)"
    n = lib.rng_circ_n(h)
    npub = lib.rng_circ_npub(h)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(npub * 4, dtype=np.uint64)
    lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
    lib.rng_circ_free(h)
    return dict(n=n, npub=npub, sel=sel, sigma=sigma, wires=wires, pubs=pubs)


class TestValidBalanceCreate:
    def test_builds_and_shape(self, vbc):
        assert vbc["npub"] == 13  # deposit(3) + commitment + recovery_id + share(8)
        assert vbc["n"] >= 1024

    def test_oracle_prove_verify(self, vbc, orc):
        n, npub = vbc["n"], vbc["npub"]
        power = max(4, int(n).bit_length())
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
        pk = o.orc_plonk_preprocess(n, npub, ptr(vbc["sel"]), ptr(vbc["sigma"]),
                                    ptr(srs_records), max_degree + 1)
        assert pk
        proof = np.zeros(157, dtype=np.uint64)
        rc = o.orc_plonk_prove(ctypes.c_void_p(pk), ptr(vbc["wires"]), ptr(vbc["pubs"]),
                               ctypes.c_uint64(3), ptr(proof))
        assert rc == 0
        tau = np.zeros(4, dtype=np.uint64)
        o.orc_derive_tau(42, ptr(tau))
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(vbc["pubs"]), ptr(proof),
                                  ptr(tau)) == 1
        # wrong statement (tamper recovery_id) -> reject
        bad = vbc["pubs"].copy()
        bad[4 * 4] ^= np.uint64(1)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(bad), ptr(proof), ptr(tau)) != 1
        o.orc_plonk_pk_free(ctypes.c_void_p(pk))

    def test_deterministic_build(self, plib):
        lib = plib.lib
        h1 = lib.rng_circ_build_vbc(7)
        h2 = lib.rng_circ_build_vbc(7)
        n = lib.rng_circ_n(h1)
        w1 = np.zeros(5 * n * 4, dtype=np.uint64)
        w2 = np.zeros(5 * n * 4, dtype=np.uint64)
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sig = np.zeros(5 * n, dtype=np.uint64)
        pubs = np.zeros(13 * 4, dtype=np.uint64)
        lib.rng_circ_get(h1, ptr(sel), ptr(sig), ptr(w1), ptr(pubs))
        lib.rng_circ_get(h2, ptr(sel), ptr(sig), ptr(w2), ptr(pubs))
        assert np.array_equal(w1, w2)
        lib.rng_circ_free(h1)
        lib.rng_circ_free(h2)
