"""Proof linking: oracle link prove/verify on the real
settlement circuit's link groups (CPU), and GPU-vs-oracle bit-exactness
(gpu-marked)."""
import ctypes

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


@pytest.fixture(scope="module")
def linkset(orc):
    from renegade_amd import load_prover
    plib = load_prover()
    lib = plib.lib
    lib.rng_circ_build_settlement.restype = ctypes.c_void_p
    lib.rng_circ_build_settlement.argtypes = [ctypes.c_uint64]
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_num_link_groups.restype = ctypes.c_uint64
    lib.rng_circ_num_link_groups.argtypes = [ctypes.c_void_p]
    lib.rng_circ_link_groups.argtypes = [ctypes.c_void_p, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]

    def tables(seed):
        h = lib.rng_circ_build_settlement(seed)
        assert h
        n = lib.rng_circ_n(h)
        npub = lib.rng_circ_npub(h)
        nlg = lib.rng_circ_num_link_groups(h)
        lg = np.zeros(3 * nlg, dtype=np.uint64)
        lib.rng_circ_link_groups(h, ptr(lg))
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sigma = np.zeros(5 * n, dtype=np.uint64)
        wires = np.zeros(5 * n * 4, dtype=np.uint64)
        pubs = np.zeros(npub * 4, dtype=np.uint64)
        lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
        lib.rng_circ_free(h)
        return dict(n=n, npub=npub, lg=lg.reshape(-1, 3), sel=sel, sigma=sigma,
                    wires=wires, pubs=pubs)

    t = tables(42)
    t2 = tables(99)  # different witness (for tamper tests)
    n, npub = t["n"], t["npub"]
    power = max(4, int(n).bit_length())
    ptau = orc.srs_generate_ptau(power, seed=42)
    max_degree = (1 << power) + 2
    g1, _, _ = orc.srs_parse(ptau, max_degree)
    srs_records = np.ascontiguousarray(g1).reshape(-1)
    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove_with_hint.argtypes = [ctypes.c_void_p, U64P, U64P,
                                            ctypes.c_uint64, U64P, U64P]
    o.orc_plonk_link.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64,
                                 ctypes.c_uint64, ctypes.c_uint64, U64P]
    o.orc_plonk_link_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P,
                                        ctypes.c_uint64, ctypes.c_uint64,
                                        ctypes.c_uint64, U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]
    pk = o.orc_plonk_preprocess(n, npub, ptr(t["sel"]), ptr(t["sigma"]),
                                ptr(srs_records), max_degree + 1)
    assert pk
    tau = np.zeros(4, dtype=np.uint64)
    o.orc_derive_tau(42, ptr(tau))

    def oprove(tbl, seed):
        proof = np.zeros(157, dtype=np.uint64)
        hint = np.zeros(4 * (n + 2) + 9, dtype=np.uint64)
        rc = o.orc_plonk_prove_with_hint(ctypes.c_void_p(pk), ptr(tbl["wires"]),
                                         ptr(tbl["pubs"]), ctypes.c_uint64(seed),
                                         ptr(proof), ptr(hint))
        assert rc == 0
        return proof, hint

    return dict(plib=plib, orc=orc, t=t, t2=t2, pk=pk, tau=tau, oprove=oprove,
                ptau=ptau, max_degree=max_degree, n=n, npub=npub)


class TestLinkOracle:
    def test_link_same_witness_different_blinding(self, linkset):
        s = linkset
        o = s["orc"].lib
        _, ha = s["oprove"](s["t"], 7)
        _, hb = s["oprove"](s["t"], 8)  # different blinders, same wire values
        for align, off, count in s["t"]["lg"]:
            lp = np.zeros(18, dtype=np.uint64)
            rc = o.orc_plonk_link(ctypes.c_void_p(s["pk"]), ptr(ha), ptr(hb),
                                  ctypes.c_uint64(int(align)), ctypes.c_uint64(int(off)),
                                  ctypes.c_uint64(int(count)), ptr(lp))
            assert rc == 0
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(s["pk"]), ptr(ha[-9:].copy()),
                                         ptr(hb[-9:].copy()), ptr(lp),
                                         ctypes.c_uint64(int(align)),
                                         ctypes.c_uint64(int(off)),
                                         ctypes.c_uint64(int(count)), ptr(s["tau"]))
            assert ok == 1, f"link group at {off} failed to verify"

    def test_link_mismatched_witness_rejected(self, linkset):
        s = linkset
        o = s["orc"].lib
        _, ha = s["oprove"](s["t"], 7)
        _, hb = s["oprove"](s["t2"], 7)  # different witness values
        align, off, count = s["t"]["lg"][0]
        lp = np.zeros(18, dtype=np.uint64)
        o.orc_plonk_link(ctypes.c_void_p(s["pk"]), ptr(ha), ptr(hb),
                         ctypes.c_uint64(int(align)), ctypes.c_uint64(int(off)),
                         ctypes.c_uint64(int(count)), ptr(lp))
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(s["pk"]), ptr(ha[-9:].copy()),
                                     ptr(hb[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(int(align)),
                                     ctypes.c_uint64(int(off)),
                                     ctypes.c_uint64(int(count)), ptr(s["tau"]))
        assert ok != 1


@pytest.mark.gpu
class TestLinkGpu:
    def test_gpu_link_bit_exact(self, linkset):
        s = linkset
        plib = s["plib"]
        if not plib.gpu_available:
            pytest.skip("no GPU")
        lib = plib.lib
        lib.rng_preprocess.restype = ctypes.c_void_p
        lib.rng_preprocess.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        lib.rng_prove.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                                  ctypes.c_uint64, U64P, U64P]
        lib.rng_link_proofs.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                                        ctypes.c_uint64, ctypes.c_uint64,
                                        ctypes.c_uint64, U64P]
        n, npub = s["n"], s["npub"]
        ctx = plib.init(s["ptau"], s["max_degree"])

        class Desc(ctypes.Structure):
            _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                        ("selectors", U64P), ("sigma", U64P),
                        ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

        t = s["t"]
        pk = lib.rng_preprocess(ctx.h, ctypes.byref(
            Desc(n, npub, ptr(t["sel"]), ptr(t["sigma"]), 0, None)))
        assert pk

        def gprove(tbl, seed):
            proof = np.zeros(157, dtype=np.uint64)
            hint = np.zeros(4 * (n + 2) + 9, dtype=np.uint64)
            assert lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(tbl["wires"]),
                                 ptr(tbl["pubs"]), seed, ptr(proof), ptr(hint)) == 0
            return proof, hint

        _, ha_g = gprove(t, 7)
        _, hb_g = gprove(t, 8)
        _, ha_o = s["oprove"](t, 7)
        _, hb_o = s["oprove"](t, 8)
        assert np.array_equal(ha_g, ha_o)  # hints bit-exact
        o = s["orc"].lib
        for align, off, count in t["lg"]:
            lp_g = np.zeros(18, dtype=np.uint64)
            assert lib.rng_link_proofs(ctx.h, ctypes.c_void_p(pk), ptr(ha_g), ptr(hb_g),
                                       int(align), int(off), int(count), ptr(lp_g)) == 0
            lp_o = np.zeros(18, dtype=np.uint64)
            assert o.orc_plonk_link(ctypes.c_void_p(s["pk"]), ptr(ha_o), ptr(hb_o),
                                    ctypes.c_uint64(int(align)),
                                    ctypes.c_uint64(int(off)),
                                    ctypes.c_uint64(int(count)), ptr(lp_o)) == 0
            assert np.array_equal(lp_g, lp_o), f"link proof mismatch at group {off}"
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(s["pk"]), ptr(ha_g[-9:].copy()),
                                         ptr(hb_g[-9:].copy()), ptr(lp_g),
                                         ctypes.c_uint64(int(align)),
                                         ctypes.c_uint64(int(off)),
                                         ctypes.c_uint64(int(count)), ptr(s["tau"]))
            assert ok == 1
        ctx.close()
