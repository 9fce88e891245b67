"""GPU prover parity: product rng_prove vs oracle prover, bit-exact
(SURVEY.md §8c contract (i)), plus oracle-verifier acceptance of GPU proofs.
"""
import ctypes

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

U64P = ctypes.POINTER(ctypes.c_uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


@pytest.fixture(scope="module")
def setup(orc):
    from renegade_amd import load_prover
    plib = load_prover()
    if not plib.gpu_available:
        pytest.skip("no GPU")
    lib = plib.lib
    lib.rng_testcirc_build.restype = ctypes.c_void_p
    lib.rng_testcirc_build.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
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
    lib.rng_pk_free.argtypes = [ctypes.c_void_p]
    lib.rng_pk_n.restype = ctypes.c_uint64
    lib.rng_pk_n.argtypes = [ctypes.c_void_p]
    lib.rng_pk_comms.argtypes = [ctypes.c_void_p, U64P]

    h = lib.rng_testcirc_build(777, 6)
    assert h
    n = lib.rng_circ_n(h)
    npub = lib.rng_circ_npub(h)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
    lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
    lib.rng_circ_free(h)

    power = max(4, int(n).bit_length())
    ptau = orc.srs_generate_ptau(power, seed=42)
    max_degree = (1 << power) + 2
    g1, _, _ = orc.srs_parse(ptau, max_degree)
    srs_records = np.ascontiguousarray(g1).reshape(-1)

    ctx = plib.init(ptau, max_degree)

    # product circuit-desc struct
    class Desc(ctypes.Structure):
        _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                    ("selectors", U64P), ("sigma", U64P),
                    ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

    desc = Desc(n, npub, ptr(sel), ptr(sigma), 0, None)
    pk = lib.rng_preprocess(ctx.h, ctypes.byref(desc))
    assert pk, "rng_preprocess failed"

    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P]
    o.orc_plonk_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]
    opk = o.orc_plonk_preprocess(n, npub, ptr(sel), ptr(sigma), ptr(srs_records),
                                 max_degree + 1)
    assert opk
    tau = np.zeros(4, dtype=np.uint64)
    o.orc_derive_tau(42, ptr(tau))
    return dict(plib=plib, lib=lib, orc=orc, ctx=ctx, pk=pk, opk=opk, n=n, npub=npub,
                wires=wires, pubs=pubs, tau=tau)


def gpu_prove(s, seed=9, with_hint=False):
    proof = np.zeros(157, dtype=np.uint64)
    hint = np.zeros(4 * (s["n"] + 2) + 9, dtype=np.uint64) if with_hint else None
    rc = s["lib"].rng_prove(s["ctx"].h, ctypes.c_void_p(s["pk"]), ptr(s["wires"]),
                            ptr(s["pubs"]), seed, ptr(proof),
                            ptr(hint) if with_hint else None)
    assert rc == 0, f"rng_prove rc={rc}"
    return (proof, hint) if with_hint else proof


class TestGpuProver:
    def test_bit_exact_vs_oracle(self, setup):
        s = setup
        gpu_proof = gpu_prove(s, seed=9)
        orc_proof = np.zeros(157, dtype=np.uint64)
        rc = s["orc"].lib.orc_plonk_prove(ctypes.c_void_p(s["opk"]), ptr(s["wires"]),
                                          ptr(s["pubs"]), ctypes.c_uint64(9),
                                          ptr(orc_proof))
        assert rc == 0
        assert np.array_equal(gpu_proof, orc_proof), \
            f"proof mismatch at {np.nonzero(gpu_proof != orc_proof)[0][:8]}"

    def test_verifies(self, setup):
        s = setup
        proof = gpu_prove(s, seed=11)
        ok = s["orc"].lib.orc_plonk_verify(ctypes.c_void_p(s["opk"]), ptr(s["pubs"]),
                                           ptr(proof), ptr(s["tau"]))
        assert ok == 1

    def test_pk_comms_match_oracle(self, setup):
        """Direct 18-commitment comparison: GPU preprocess vs oracle
        preprocess on the same tables/SRS (VERDICT r01 missing #4)."""
        s = setup
        comms = np.zeros(18 * 9, dtype=np.uint64)
        s["lib"].rng_pk_comms(ctypes.c_void_p(s["pk"]), ptr(comms))
        ocomms = np.zeros(18 * 9, dtype=np.uint64)
        o = s["orc"].lib
        o.orc_plonk_pk_comms.argtypes = [ctypes.c_void_p, U64P]
        o.orc_plonk_pk_comms(ctypes.c_void_p(s["opk"]), ptr(ocomms))
        assert np.array_equal(comms, ocomms), \
            f"PK commitment mismatch at records {set((np.nonzero(comms != ocomms)[0] // 9).tolist())}"
        assert np.any(comms != 0)

    def test_cohort_bit_exact_vs_single(self, setup):
        """rng_prove_cohort(k) proofs (and hints) are bit-identical to k
        independent rng_prove calls with the same seeds."""
        s = setup
        lib = s["lib"]
        n, npub = int(s["n"]), int(s["npub"])
        k = 5
        lib.rng_prove_cohort.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_uint64, U64P, U64P, U64P, U64P, U64P]
        wires_all = np.tile(s["wires"], k)
        pubs_all = np.tile(s["pubs"], k)
        seeds = np.array([31, 32, 33, 31, 99], dtype=np.uint64)
        proofs = np.zeros(157 * k, dtype=np.uint64)
        hint_len = 4 * (n + 2) + 9
        hints = np.zeros(hint_len * k, dtype=np.uint64)
        rc = lib.rng_prove_cohort(s["ctx"].h, ctypes.c_void_p(s["pk"]), k,
                                  ptr(wires_all), ptr(pubs_all), ptr(seeds),
                                  ptr(proofs), ptr(hints))
        assert rc == 0, f"rng_prove_cohort rc={rc}"
        for p in range(k):
            single, shint = gpu_prove(s, seed=int(seeds[p]), with_hint=True)
            assert np.array_equal(proofs[157 * p:157 * (p + 1)], single), \
                f"cohort proof {p} differs from single-proof path"
            assert np.array_equal(hints[hint_len * p:hint_len * (p + 1)], shint), \
                f"cohort hint {p} differs"
        # same seed -> same proof within a cohort too
        assert np.array_equal(proofs[:157], proofs[157 * 3:157 * 4])

    def test_link_hint(self, setup):
        s = setup
        proof, hint = gpu_prove(s, seed=13, with_hint=True)
        # hint commitment equals the first wire commitment in the proof
        assert np.array_equal(hint[-9:], proof[:9])

    def test_seed_determinism(self, setup):
        s = setup
        p1 = gpu_prove(s, seed=21)
        p2 = gpu_prove(s, seed=21)
        p3 = gpu_prove(s, seed=22)
        assert np.array_equal(p1, p2)
        assert not np.array_equal(p1, p3)


@pytest.mark.parametrize("builder,seed", [("rng_circ_build_settlement", 42),
                                          ("rng_circ_build_vbc", 42),
                                          ("rng_circ_build_valid_deposit", 42),
                                          ("rng_circ_build_valid_withdrawal", 42),
                                          ("rng_circ_build_validity", 42),
                                          ("rng_circ_build_ob_validity", 42),
                                          ("rng_circ_build_valid_order_cancellation", 42),
                                          ("rng_circ_build_public_settlement", 42),
                                          ("rng_circ_build_io_settlement", 42),
                                          ("rng_circ_build_io_validity", 42),
                                          ("rng_circ_build_io_bounded_settlement", 42),
                                          ("rng_circ_build_ib_bounded_settlement", 42),
                                          ("rng_circ_build_note_redemption", 42),
                                          ("rng_circ_build_fee_public_relayer", 42),
                                          ("rng_circ_build_fee_public_protocol", 42),
                                          ("rng_circ_build_fee_private_relayer", 42),
                                          ("rng_circ_build_ioff", 42),
                                          ("rng_circ_build_ff_validity", 42),
                                          ("rng_circ_build_nob_validity", 42),
                                          ("rng_circ_build_fee_private_protocol", 42)])
def test_real_circuit_gpu_parity(orc, builder, seed):
    """GPU prover bit-exact vs oracle on the REAL circuits (settlement =
    BASELINE config #4, VBC = config #1)."""
    from renegade_amd import load_prover
    plib = load_prover()
    if not plib.gpu_available:
        pytest.skip("no GPU")
    lib = plib.lib
    fn = getattr(lib, builder)
    fn.restype = ctypes.c_void_p
    two_arg = builder in ("rng_circ_build_validity", "rng_circ_build_ob_validity",
                          "rng_circ_build_ff_validity")  # (seed, party)
    fn.argtypes = [ctypes.c_uint64, ctypes.c_uint64] if two_arg else [ctypes.c_uint64]
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
    h = fn(seed, 0) if two_arg else fn(seed)
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
    ctx = plib.init(ptau, max_degree)

    class Desc(ctypes.Structure):
        _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                    ("selectors", U64P), ("sigma", U64P),
                    ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

    pk = lib.rng_preprocess(ctx.h, ctypes.byref(Desc(n, npub, ptr(sel), ptr(sigma), 0,
                                                     None)))
    assert pk
    proof = np.zeros(157, dtype=np.uint64)
    assert lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(wires), ptr(pubs), 7,
                         ptr(proof), None) == 0
    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P]
    o.orc_plonk_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]
    opk = o.orc_plonk_preprocess(n, npub, ptr(sel), ptr(sigma), ptr(srs_records),
                                 max_degree + 1)
    orc_proof = np.zeros(157, dtype=np.uint64)
    assert o.orc_plonk_prove(ctypes.c_void_p(opk), ptr(wires), ptr(pubs),
                             ctypes.c_uint64(7), ptr(orc_proof)) == 0
    assert np.array_equal(proof, orc_proof)
    tau = np.zeros(4, dtype=np.uint64)
    o.orc_derive_tau(42, ptr(tau))
    assert o.orc_plonk_verify(ctypes.c_void_p(opk), ptr(pubs), ptr(proof), ptr(tau)) == 1
    # product verifier (real pairing, traits.rs:1012 replacement)
    lib.rng_verify.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P]
    lib.rng_verify.restype = ctypes.c_int
    assert lib.rng_verify(ctx.h, ctypes.c_void_p(pk), ptr(pubs), ptr(proof)) == 0
    bad = proof.copy()
    bad[118] ^= np.uint64(1)  # tamper a wire eval
    assert lib.rng_verify(ctx.h, ctypes.c_void_p(pk), ptr(pubs), ptr(bad)) != 0
    badp = pubs.copy()
    badp[0] ^= np.uint64(1)
    assert lib.rng_verify(ctx.h, ctypes.c_void_p(pk), ptr(badp), ptr(proof)) != 0
    lib.rng_pk_free(ctypes.c_void_p(pk))
    ctx.close()


def test_cohort_large_domain_bit_exact(orc):
    """Cohort parity at a production validity domain (n=16384): the bundle
    bench cohorts these tables, so pin them against the single-proof path
    (the small-n cohort test exercises a different NTT/MSM shape)."""
    from renegade_amd import load_prover
    plib = load_prover()
    if not plib.gpu_available:
        pytest.skip("no GPU")
    lib = plib.lib
    lib.rng_circ_build_validity.restype = ctypes.c_void_p
    lib.rng_circ_build_validity.argtypes = [ctypes.c_uint64] * 2
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
    h = lib.rng_circ_build_validity(11, 0)
    assert h
    n = lib.rng_circ_n(h)
    assert int(n) >= 16384
    npub = lib.rng_circ_npub(h)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
    lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
    lib.rng_circ_free(h)
    power = max(4, int(n).bit_length())
    from tests.orc_bindings import OracleLib  # noqa: F811 — reuse orc fixture lib
    ptau = orc.srs_generate_ptau(power, seed=42)
    ctx = plib.init(ptau, (1 << power) + 2)

    class Desc(ctypes.Structure):
        _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                    ("selectors", U64P), ("sigma", U64P),
                    ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

    pk = lib.rng_preprocess(ctx.h, ctypes.byref(Desc(n, npub, ptr(sel), ptr(sigma),
                                                     0, None)))
    assert pk
    k = 3
    seeds = np.array([71, 72, 71], dtype=np.uint64)
    proofs = np.zeros(157 * k, dtype=np.uint64)
    hints = np.zeros((4 * (int(n) + 2) + 9) * k, dtype=np.uint64)
    rc = lib.rng_prove_cohort(ctx.h, ctypes.c_void_p(pk), k, ptr(np.tile(wires, k)),
                              ptr(np.tile(pubs, k)), ptr(seeds), ptr(proofs),
                              ptr(hints))
    assert rc == 0
    single = np.zeros(157, dtype=np.uint64)
    for p in range(k):
        assert lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(wires), ptr(pubs),
                             ctypes.c_uint64(int(seeds[p])), ptr(single), None) == 0
        assert np.array_equal(single, proofs[157 * p:157 * (p + 1)]), \
            f"large-domain cohort proof {p} differs"
    ctx.close()
