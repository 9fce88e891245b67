"""Cross-domain proof linking: two circuits with DIFFERENT domain sizes share
a link group on the alignment grid (mpc-relation GroupLayout semantics) —
the production shape of validity-proof <-> settlement-proof links
(proof_linking/intent_and_balance.rs).  CPU via the oracle."""
import ctypes

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

ALIGN, OFF, COUNT = 6, 1, 17


@pytest.fixture(scope="module")
def xd(orc):
    from renegade_amd import load_prover
    plib = load_prover()
    lib = plib.lib
    lib.rng_testcirc_build_linked.restype = ctypes.c_void_p
    lib.rng_testcirc_build_linked.argtypes = [ctypes.c_uint64] * 5
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]

    def tables(value_seed, scale):
        h = lib.rng_testcirc_build_linked(value_seed, scale, ALIGN, OFF, COUNT)
        assert h
        n = lib.rng_circ_n(h)
        npub = lib.rng_circ_npub(h)
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sigma = np.zeros(5 * n, dtype=np.uint64)
        wires = np.zeros(5 * n * 4, dtype=np.uint64)
        pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
        lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
        lib.rng_circ_free(h)
        return dict(n=n, npub=npub, sel=sel, sigma=sigma, wires=wires, pubs=pubs)

    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove_with_hint.argtypes = [ctypes.c_void_p, U64P, U64P,
                                            ctypes.c_uint64, U64P, U64P]
    o.orc_plonk_link.argtypes = [ctypes.c_void_p, U64P, U64P] + [ctypes.c_uint64] * 3 + [U64P]
    o.orc_plonk_link_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P] + \
        [ctypes.c_uint64] * 3 + [U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]

    def setup_circuit(t):
        power = max(4, int(t["n"]).bit_length())
        ptau = orc.srs_generate_ptau(power, seed=42)
        md = (1 << power) + 2
        g1, _, _ = orc.srs_parse(ptau, md)
        srs = np.ascontiguousarray(g1).reshape(-1)
        pk = o.orc_plonk_preprocess(t["n"], t["npub"], ptr(t["sel"]), ptr(t["sigma"]),
                                    ptr(srs), md + 1)
        assert pk
        return pk

    def prove(pk, t, seed):
        proof = np.zeros(157, dtype=np.uint64)
        hint = np.zeros(4 * (t["n"] + 2) + 9, dtype=np.uint64)
        assert o.orc_plonk_prove_with_hint(ctypes.c_void_p(pk), ptr(t["wires"]),
                                           ptr(t["pubs"]), ctypes.c_uint64(seed),
                                           ptr(proof), ptr(hint)) == 0
        return proof, hint

    tau = np.zeros(4, dtype=np.uint64)
    o.orc_derive_tau(42, ptr(tau))
    return dict(orc=orc, tables=tables, setup=setup_circuit, prove=prove, tau=tau, o=o)


def _truncate_hint(hint, n_small):
    """Link hints carry (n+2) coefficients; the verifier/link only needs the
    polynomial — pass each side's own full hint."""
    return hint


class TestCrossDomainLink:
    def test_different_domains_link(self, xd):
        o = xd["o"]
        tA = xd["tables"](7, 2)    # small circuit
        tB = xd["tables"](7, 30)   # much larger circuit, same link values
        assert tA["n"] != tB["n"], "need distinct domain sizes"
        pkA = xd["setup"](tA)
        pkB = xd["setup"](tB)
        _, hA = xd["prove"](pkA, tA, 3)
        _, hB = xd["prove"](pkB, tB, 4)
        # hints have different lengths; orc_plonk_link takes same-pk pairs, so
        # link via the LARGER pk with B's hint and A's hint zero-extended
        nB = tB["n"]
        hA_ext = np.zeros(4 * (nB + 2) + 9, dtype=np.uint64)
        hA_ext[:4 * (tA["n"] + 2)] = hA[:4 * (tA["n"] + 2)]
        hA_ext[-9:] = hA[-9:]
        lp = np.zeros(18, dtype=np.uint64)
        assert o.orc_plonk_link(ctypes.c_void_p(pkB), ptr(hA_ext), ptr(hB),
                                ctypes.c_uint64(ALIGN), ctypes.c_uint64(OFF),
                                ctypes.c_uint64(COUNT), ptr(lp)) == 0
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pkB), ptr(hA_ext[-9:].copy()),
                                     ptr(hB[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(ALIGN), ctypes.c_uint64(OFF),
                                     ctypes.c_uint64(COUNT), ptr(xd["tau"]))
        assert ok == 1, "cross-domain link failed"

    def test_different_values_rejected(self, xd):
        o = xd["o"]
        tA = xd["tables"](7, 2)
        tB = xd["tables"](8, 30)   # DIFFERENT link values
        pkA = xd["setup"](tA)
        pkB = xd["setup"](tB)
        _, hA = xd["prove"](pkA, tA, 3)
        _, hB = xd["prove"](pkB, tB, 4)
        nB = tB["n"]
        hA_ext = np.zeros(4 * (nB + 2) + 9, dtype=np.uint64)
        hA_ext[:4 * (tA["n"] + 2)] = hA[:4 * (tA["n"] + 2)]
        hA_ext[-9:] = hA[-9:]
        lp = np.zeros(18, dtype=np.uint64)
        o.orc_plonk_link(ctypes.c_void_p(pkB), ptr(hA_ext), ptr(hB),
                         ctypes.c_uint64(ALIGN), ctypes.c_uint64(OFF),
                         ctypes.c_uint64(COUNT), ptr(lp))
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pkB), ptr(hA_ext[-9:].copy()),
                                     ptr(hB[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(ALIGN), ctypes.c_uint64(OFF),
                                     ctypes.c_uint64(COUNT), ptr(xd["tau"]))
        assert ok != 1


    def test_wrong_geometry_rejected(self, xd):
        """A link proof computed at the right placement must NOT verify at a
        different offset, alignment or count."""
        o = xd["o"]
        tA = xd["tables"](7, 2)
        tB = xd["tables"](7, 30)
        pkB = xd["setup"](tB)
        _, hA = xd["prove"](xd["setup"](tA), tA, 3)
        _, hB = xd["prove"](pkB, tB, 4)
        nB = tB["n"]
        hA_ext = np.zeros(4 * (nB + 2) + 9, dtype=np.uint64)
        hA_ext[:4 * (tA["n"] + 2)] = hA[:4 * (tA["n"] + 2)]
        hA_ext[-9:] = hA[-9:]
        lp = np.zeros(18, dtype=np.uint64)
        assert o.orc_plonk_link(ctypes.c_void_p(pkB), ptr(hA_ext), ptr(hB),
                                ctypes.c_uint64(ALIGN), ctypes.c_uint64(OFF),
                                ctypes.c_uint64(COUNT), ptr(lp)) == 0
        for a, off, cnt in [(ALIGN, OFF + 1, COUNT),   # wrong offset
                            (ALIGN + 1, OFF, COUNT),   # wrong alignment
                            (ALIGN, OFF, COUNT - 1)]:  # wrong count
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(pkB), ptr(hA_ext[-9:].copy()),
                                         ptr(hB[-9:].copy()), ptr(lp),
                                         ctypes.c_uint64(a), ctypes.c_uint64(off),
                                         ctypes.c_uint64(cnt), ptr(xd["tau"]))
            assert ok != 1, f"verified at wrong geometry {(a, off, cnt)}"
