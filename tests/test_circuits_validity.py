"""`INTENT AND BALANCE VALIDITY` circuit
(zk_circuits/validity_proofs/intent_and_balance.rs) — build, shape, oracle
prove/verify, and the PRODUCTION cross-domain proof link: the validity
proof's party group links into the settlement proof of the same bundle
(proof_linking_groups inherits the settlement layout, :316-341)."""
import ctypes

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


@pytest.fixture(scope="module")
def vb(orc):
    from renegade_amd import load_prover
    plib = load_prover()
    lib = plib.lib
    lib.rng_circ_build_validity.restype = ctypes.c_void_p
    lib.rng_circ_build_validity.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
    lib.rng_circ_build_ob_validity.restype = ctypes.c_void_p
    lib.rng_circ_build_ob_validity.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
    lib.rng_circ_build_settlement_bundle.restype = ctypes.c_void_p
    lib.rng_circ_build_settlement_bundle.argtypes = [ctypes.c_uint64]
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_num_link_groups.restype = ctypes.c_uint64
    lib.rng_circ_num_link_groups.argtypes = [ctypes.c_void_p]
    lib.rng_circ_link_groups.argtypes = [ctypes.c_void_p, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]

    def fetch(h):
        assert h
        n = lib.rng_circ_n(h)
        npub = lib.rng_circ_npub(h)
        nlg = lib.rng_circ_num_link_groups(h)
        lg = np.zeros(3 * max(1, nlg), dtype=np.uint64)
        lib.rng_circ_link_groups(h, ptr(lg))
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sigma = np.zeros(5 * n, dtype=np.uint64)
        wires = np.zeros(5 * n * 4, dtype=np.uint64)
        pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
        lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
        lib.rng_circ_free(h)
        return dict(n=n, npub=npub, nlg=nlg, lg=lg.reshape(-1, 3), sel=sel,
                    sigma=sigma, wires=wires, pubs=pubs)

    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove_with_hint.argtypes = [ctypes.c_void_p, U64P, U64P,
                                            ctypes.c_uint64, U64P, U64P]
    o.orc_plonk_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P]
    o.orc_plonk_link.argtypes = [ctypes.c_void_p, U64P, U64P] + \
        [ctypes.c_uint64] * 3 + [U64P]
    o.orc_plonk_link_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P] + \
        [ctypes.c_uint64] * 3 + [U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]

    def setup(t):
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
    return dict(lib=lib, o=o, fetch=fetch, setup=setup, prove=prove, tau=tau)


class TestValidityCircuit:
    def test_shape(self, vb):
        lib = vb["lib"]
        t = vb["fetch"](lib.rng_circ_build_validity(42, 0))
        # statement = 10 scalars (intent_and_balance.rs:277-309)
        assert t["npub"] == 10
        # two link groups, both 17 vars, inheriting the settlement placement
        assert t["nlg"] == 2
        ts = vb["fetch"](lib.rng_circ_build_settlement_bundle(42))
        stl_groups = {tuple(int(x) for x in row) for row in ts["lg"]}
        for row in t["lg"]:
            assert int(row[2]) == 17
            assert tuple(int(x) for x in row) in stl_groups, \
                "validity group not at the settlement's layout"
        assert t["n"] != ts["n"], "expect distinct domain sizes (cross-domain link)"

    def test_seeds_vary(self, vb):
        lib = vb["lib"]
        for seed in [1, 2, 3]:
            for party in (0, 1):
                h = lib.rng_circ_build_validity(seed, party)
                assert h, f"seed {seed} party {party} unsatisfied"
                lib.rng_circ_free(h)

    def test_oracle_prove_verify(self, vb):
        lib, o = vb["lib"], vb["o"]
        t = vb["fetch"](lib.rng_circ_build_validity(42, 0))
        pk = vb["setup"](t)
        proof, _ = vb["prove"](pk, t, 11)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(t["pubs"]), ptr(proof),
                                  ptr(vb["tau"])) == 1
        # tamper the intent nullifier -> reject
        bad = t["pubs"].copy()
        bad[1 * 4] ^= np.uint64(1)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(bad), ptr(proof),
                                  ptr(vb["tau"])) != 1
        o.orc_plonk_pk_free(ctypes.c_void_p(pk))


class TestObValidityCircuit:
    def test_shape_and_prove(self, vb):
        lib, o = vb["lib"], vb["o"]
        t = vb["fetch"](lib.rng_circ_build_ob_validity(42, 0))
        # statement = 5 scalars (output_balance.rs:163-176)
        assert t["npub"] == 5
        # both out-groups, 11 vars, at the settlement's layout
        assert t["nlg"] == 2
        ts = vb["fetch"](lib.rng_circ_build_settlement_bundle(42))
        stl_groups = {tuple(int(x) for x in row) for row in ts["lg"]}
        for row in t["lg"]:
            assert int(row[2]) == 11
            assert tuple(int(x) for x in row) in stl_groups
        pk = vb["setup"](t)
        proof, _ = vb["prove"](pk, t, 13)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(t["pubs"]), ptr(proof),
                                  ptr(vb["tau"])) == 1
        bad = t["pubs"].copy()
        bad[4 * 4] ^= np.uint64(1)  # recovery id
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(bad), ptr(proof),
                                  ptr(vb["tau"])) != 1
        o.orc_plonk_pk_free(ctypes.c_void_p(pk))


def bundle_groups(ts_lg):
    """Settlement link groups by (kind, party): count-17 rows are the
    intent_and_balance party groups, count-11 the output_balance groups;
    within a kind, party0 has the smaller grid offset (creation order)."""
    party = sorted((int(r[1]), (int(r[0]), int(r[1]), int(r[2])))
                   for r in ts_lg if int(r[2]) == 17)
    out = sorted((int(r[1]), (int(r[0]), int(r[1]), int(r[2])))
                 for r in ts_lg if int(r[2]) == 11)
    return {("party", 0): party[0][1], ("party", 1): party[1][1],
            ("out", 0): out[0][1], ("out", 1): out[1][1]}


class TestBundleLink:
    """The production proof bundle (native_proof_manager.rs:554-590):
    settlement proof + 4 validity proofs, linked across DIFFERENT domain
    sizes."""

    def test_full_bundle_links(self, vb):
        lib, o = vb["lib"], vb["o"]
        ts = vb["fetch"](lib.rng_circ_build_settlement_bundle(7))
        t0 = vb["fetch"](lib.rng_circ_build_validity(7, 0))
        t1 = vb["fetch"](lib.rng_circ_build_validity(7, 1))
        u0 = vb["fetch"](lib.rng_circ_build_ob_validity(7, 0))
        u1 = vb["fetch"](lib.rng_circ_build_ob_validity(7, 1))
        pk_s = vb["setup"](ts)
        pk_v = vb["setup"](t0)  # same shape for both parties
        pk_u = vb["setup"](u0)
        _, hs = vb["prove"](pk_s, ts, 3)
        _, h0 = vb["prove"](pk_v, t0, 4)
        _, h1 = vb["prove"](pk_v, t1, 5)
        _, g0 = vb["prove"](pk_u, u0, 6)
        _, g1 = vb["prove"](pk_u, u1, 7)
        n_big = max(int(ts["n"]), int(t0["n"]), int(u0["n"]))
        pk_big = {int(ts["n"]): pk_s, int(t0["n"]): pk_v, int(u0["n"]): pk_u}[n_big]

        def ext(h, n_small):
            out = np.zeros(4 * (n_big + 2) + 9, dtype=np.uint64)
            out[:4 * (n_small + 2)] = h[:4 * (n_small + 2)]
            out[-9:] = h[-9:]
            return out

        hs_e = ext(hs, int(ts["n"]))
        groups = bundle_groups(ts["lg"])
        legs = [(("party", 0), ext(h0, int(t0["n"]))),
                (("party", 1), ext(h1, int(t1["n"]))),
                (("out", 0), ext(g0, int(u0["n"]))),
                (("out", 1), ext(g1, int(u1["n"])))]
        for key, hv_e in legs:
            a, off, cnt = groups[key]
            lp = np.zeros(18, dtype=np.uint64)
            rc = o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hv_e), ptr(hs_e),
                                  ctypes.c_uint64(a), ctypes.c_uint64(off),
                                  ctypes.c_uint64(cnt), ptr(lp))
            assert rc == 0
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big),
                                         ptr(hv_e[-9:].copy()), ptr(hs_e[-9:].copy()),
                                         ptr(lp), ctypes.c_uint64(a),
                                         ctypes.c_uint64(off), ctypes.c_uint64(cnt),
                                         ptr(vb["tau"]))
            assert ok == 1, f"{key} <-> settlement link failed"

        # negative: party0's validity proof must NOT link at party1's offset
        a, off, cnt = groups[("party", 1)]
        h0_e = ext(h0, int(t0["n"]))
        lp = np.zeros(18, dtype=np.uint64)
        o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(h0_e), ptr(hs_e),
                         ctypes.c_uint64(a), ctypes.c_uint64(off),
                         ctypes.c_uint64(cnt), ptr(lp))
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big), ptr(h0_e[-9:].copy()),
                                     ptr(hs_e[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(a), ctypes.c_uint64(off),
                                     ctypes.c_uint64(cnt), ptr(vb["tau"]))
        assert ok != 1, "cross-party link should not verify"

        # --- public settlement: same validity proofs link into the
        #     single-party public settlement at the party-0 layouts ---
        lib.rng_circ_build_public_settlement.restype = ctypes.c_void_p
        lib.rng_circ_build_public_settlement.argtypes = [ctypes.c_uint64]
        tp = vb["fetch"](lib.rng_circ_build_public_settlement(7))
        assert tp["npub"] == 14
        pk_p = vb["setup"](tp)
        _, hp = vb["prove"](pk_p, tp, 8)
        hp_e = ext(hp, int(tp["n"]))
        for key, hv_e in legs[:1] + legs[2:3]:  # (party,0) and (out,0)
            a, off, cnt = groups[key]
            lp = np.zeros(18, dtype=np.uint64)
            assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hv_e), ptr(hp_e),
                                    ctypes.c_uint64(a), ctypes.c_uint64(off),
                                    ctypes.c_uint64(cnt), ptr(lp)) == 0
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big),
                                         ptr(hv_e[-9:].copy()), ptr(hp_e[-9:].copy()),
                                         ptr(lp), ctypes.c_uint64(a),
                                         ctypes.c_uint64(off), ctypes.c_uint64(cnt),
                                         ptr(vb["tau"]))
            assert ok == 1, f"{key} <-> public settlement link failed"
        o.orc_plonk_pk_free(ctypes.c_void_p(pk_p))

        # --- bounded settlement: same validity proofs link into the
        #     bounded variant at the party-0 layouts ---
        lib.rng_circ_build_ib_bounded_settlement.restype = ctypes.c_void_p
        lib.rng_circ_build_ib_bounded_settlement.argtypes = [ctypes.c_uint64]
        tbb = vb["fetch"](lib.rng_circ_build_ib_bounded_settlement(7))
        assert tbb["npub"] == 16
        pk_bb = vb["setup"](tbb)
        _, hbb = vb["prove"](pk_bb, tbb, 9)
        hbb_e = ext(hbb, int(tbb["n"]))
        for key, hv_e in legs[:1] + legs[2:3]:  # (party,0) and (out,0)
            a, off, cnt = groups[key]
            lp = np.zeros(18, dtype=np.uint64)
            assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hv_e), ptr(hbb_e),
                                    ctypes.c_uint64(a), ctypes.c_uint64(off),
                                    ctypes.c_uint64(cnt), ptr(lp)) == 0
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big),
                                         ptr(hv_e[-9:].copy()), ptr(hbb_e[-9:].copy()),
                                         ptr(lp), ctypes.c_uint64(a),
                                         ctypes.c_uint64(off), ctypes.c_uint64(cnt),
                                         ptr(vb["tau"]))
            assert ok == 1, f"{key} <-> bounded settlement link failed"
        o.orc_plonk_pk_free(ctypes.c_void_p(pk_bb))
        for pk in (pk_s, pk_v, pk_u):
            o.orc_plonk_pk_free(ctypes.c_void_p(pk))


class TestIntentOnly:
    """INTENT ONLY VALIDITY <-> INTENT ONLY PUBLIC SETTLEMENT
    (validity_proofs/intent_only.rs, intent_only_public_settlement.rs):
    one 5-var link group placed by the settlement, inherited cross-domain."""

    def test_pair_and_link(self, vb):
        lib, o = vb["lib"], vb["o"]
        lib.rng_circ_build_io_settlement.restype = ctypes.c_void_p
        lib.rng_circ_build_io_settlement.argtypes = [ctypes.c_uint64]
        lib.rng_circ_build_io_validity.restype = ctypes.c_void_p
        lib.rng_circ_build_io_validity.argtypes = [ctypes.c_uint64]
        ts = vb["fetch"](lib.rng_circ_build_io_settlement(5))
        tv = vb["fetch"](lib.rng_circ_build_io_validity(5))
        assert ts["npub"] == 6 and tv["npub"] == 7
        assert [tuple(map(int, r)) for r in ts["lg"]] == \
            [tuple(map(int, r)) for r in tv["lg"]]
        pk_s = vb["setup"](ts)
        pk_v = vb["setup"](tv)
        ps, hs = vb["prove"](pk_s, ts, 3)
        pv_, hv = vb["prove"](pk_v, tv, 4)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk_s), ptr(ts["pubs"]), ptr(ps),
                                  ptr(vb["tau"])) == 1
        assert o.orc_plonk_verify(ctypes.c_void_p(pk_v), ptr(tv["pubs"]), ptr(pv_),
                                  ptr(vb["tau"])) == 1
        n_big = max(int(ts["n"]), int(tv["n"]))
        pk_big = pk_v if int(tv["n"]) == n_big else pk_s

        def ext(h, n_small):
            out = np.zeros(4 * (n_big + 2) + 9, dtype=np.uint64)
            out[:4 * (n_small + 2)] = h[:4 * (n_small + 2)]
            out[-9:] = h[-9:]
            return out

        a, off, cnt = (int(x) for x in tv["lg"][0])
        hv_e, hs_e = ext(hv, int(tv["n"])), ext(hs, int(ts["n"]))
        lp = np.zeros(18, dtype=np.uint64)
        assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hv_e), ptr(hs_e),
                                ctypes.c_uint64(a), ctypes.c_uint64(off),
                                ctypes.c_uint64(cnt), ptr(lp)) == 0
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big), ptr(hv_e[-9:].copy()),
                                     ptr(hs_e[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(a), ctypes.c_uint64(off),
                                     ctypes.c_uint64(cnt), ptr(vb["tau"]))
        assert ok == 1, "intent-only validity <-> settlement link failed"

        # bounded settlement (intent_only_bounded_settlement.rs): same intent
        # links into the bounded variant at the same inherited layout
        lib.rng_circ_build_io_bounded_settlement.restype = ctypes.c_void_p
        lib.rng_circ_build_io_bounded_settlement.argtypes = [ctypes.c_uint64]
        tb = vb["fetch"](lib.rng_circ_build_io_bounded_settlement(5))
        assert tb["npub"] == 9
        pk_b = vb["setup"](tb)
        pb, hb = vb["prove"](pk_b, tb, 9)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk_b), ptr(tb["pubs"]), ptr(pb),
                                  ptr(vb["tau"])) == 1
        hb_e = ext(hb, int(tb["n"]))
        lp2 = np.zeros(18, dtype=np.uint64)
        assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hv_e), ptr(hb_e),
                                ctypes.c_uint64(a), ctypes.c_uint64(off),
                                ctypes.c_uint64(cnt), ptr(lp2)) == 0
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big), ptr(hv_e[-9:].copy()),
                                     ptr(hb_e[-9:].copy()), ptr(lp2),
                                     ctypes.c_uint64(a), ctypes.c_uint64(off),
                                     ctypes.c_uint64(cnt), ptr(vb["tau"]))
        assert ok == 1, "intent-only validity <-> bounded settlement link failed"
        o.orc_plonk_pk_free(ctypes.c_void_p(pk_b))

        # first-fill validity (intent_only_first_fill.rs): same intent links
        # into the settlement from a first-fill proof as well
        lib.rng_circ_build_ioff.restype = ctypes.c_void_p
        lib.rng_circ_build_ioff.argtypes = [ctypes.c_uint64]
        tf = vb["fetch"](lib.rng_circ_build_ioff(5))
        assert tf["npub"] == 8
        pk_f = vb["setup"](tf)
        pf, hf = vb["prove"](pk_f, tf, 10)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk_f), ptr(tf["pubs"]), ptr(pf),
                                  ptr(vb["tau"])) == 1
        hf_e = ext(hf, int(tf["n"]))
        lp3 = np.zeros(18, dtype=np.uint64)
        assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hf_e), ptr(hs_e),
                                ctypes.c_uint64(a), ctypes.c_uint64(off),
                                ctypes.c_uint64(cnt), ptr(lp3)) == 0
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big), ptr(hf_e[-9:].copy()),
                                     ptr(hs_e[-9:].copy()), ptr(lp3),
                                     ctypes.c_uint64(a), ctypes.c_uint64(off),
                                     ctypes.c_uint64(cnt), ptr(vb["tau"]))
        assert ok == 1, "first-fill validity <-> settlement link failed"
        o.orc_plonk_pk_free(ctypes.c_void_p(pk_f))

        # a settlement over a DIFFERENT intent must not link
        ts2 = vb["fetch"](lib.rng_circ_build_io_settlement(6))
        pk_s2 = vb["setup"](ts2)
        _, hs2 = vb["prove"](pk_s2, ts2, 3)
        hs2_e = ext(hs2, int(ts2["n"]))
        o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hv_e), ptr(hs2_e),
                         ctypes.c_uint64(a), ctypes.c_uint64(off),
                         ctypes.c_uint64(cnt), ptr(lp))
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big), ptr(hv_e[-9:].copy()),
                                     ptr(hs2_e[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(a), ctypes.c_uint64(off),
                                     ctypes.c_uint64(cnt), ptr(vb["tau"]))
        assert ok != 1
        for pk in (pk_s, pk_v, pk_s2):
            o.orc_plonk_pk_free(ctypes.c_void_p(pk))


class TestSchnorrCircuits:
    """The three embedded-curve circuits (Schnorr / ElGamal): build +
    satisfiability (the builders return null on an unsatisfied circuit), and
    the first-fill proof linking into the settlement bundle."""

    def test_builders_satisfied(self, vb):
        lib = vb["lib"]
        lib.rng_circ_build_fee_private_protocol.restype = ctypes.c_void_p
        lib.rng_circ_build_fee_private_protocol.argtypes = [ctypes.c_uint64]
        lib.rng_circ_build_nob_validity.restype = ctypes.c_void_p
        lib.rng_circ_build_nob_validity.argtypes = [ctypes.c_uint64]
        lib.rng_circ_build_ff_validity.restype = ctypes.c_void_p
        lib.rng_circ_build_ff_validity.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
        for h, what in [(lib.rng_circ_build_fee_private_protocol(3), "fee pp"),
                        (lib.rng_circ_build_nob_validity(3), "nob"),
                        (lib.rng_circ_build_ff_validity(3, 1), "ff p1")]:
            assert h, f"{what} unsatisfied"
            lib.rng_circ_free(h)

    def test_schnorr_gadget_tamper(self, vb):
        lib = vb["lib"]
        lib.rng_testcirc_schnorr.restype = ctypes.c_void_p
        lib.rng_testcirc_schnorr.argtypes = [ctypes.c_uint64, ctypes.c_int]
        lib.rng_testcirc_elgamal.restype = ctypes.c_void_p
        lib.rng_testcirc_elgamal.argtypes = [ctypes.c_uint64, ctypes.c_int]
        h = lib.rng_testcirc_schnorr(9, 0)
        assert h
        lib.rng_circ_free(h)
        assert not lib.rng_testcirc_schnorr(9, 1), "tampered signature accepted"
        h = lib.rng_testcirc_elgamal(9, 0)
        assert h
        lib.rng_circ_free(h)
        assert not lib.rng_testcirc_elgamal(9, 1), "tampered ciphertext accepted"

    def test_ff_links_into_settlement(self, vb):
        lib, o = vb["lib"], vb["o"]
        lib.rng_circ_build_ff_validity.restype = ctypes.c_void_p
        lib.rng_circ_build_ff_validity.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
        ts = vb["fetch"](lib.rng_circ_build_settlement_bundle(7))
        tf = vb["fetch"](lib.rng_circ_build_ff_validity(7, 0))
        assert tf["npub"] == 11
        pk_s = vb["setup"](ts)
        pk_f = vb["setup"](tf)
        _, hs = vb["prove"](pk_s, ts, 3)
        pf, hf = vb["prove"](pk_f, tf, 4)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk_f), ptr(tf["pubs"]), ptr(pf),
                                  ptr(vb["tau"])) == 1
        n_big = max(int(ts["n"]), int(tf["n"]))
        pk_big = pk_f if int(tf["n"]) == n_big else pk_s

        def ext(h, n_small):
            out = np.zeros(4 * (n_big + 2) + 9, dtype=np.uint64)
            out[:4 * (n_small + 2)] = h[:4 * (n_small + 2)]
            out[-9:] = h[-9:]
            return out

        groups = bundle_groups(ts["lg"])
        a, off, cnt = groups[("party", 0)]
        hf_e, hs_e = ext(hf, int(tf["n"])), ext(hs, int(ts["n"]))
        lp = np.zeros(18, dtype=np.uint64)
        assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(hf_e), ptr(hs_e),
                                ctypes.c_uint64(a), ctypes.c_uint64(off),
                                ctypes.c_uint64(cnt), ptr(lp)) == 0
        ok = o.orc_plonk_link_verify(ctypes.c_void_p(pk_big), ptr(hf_e[-9:].copy()),
                                     ptr(hs_e[-9:].copy()), ptr(lp),
                                     ctypes.c_uint64(a), ctypes.c_uint64(off),
                                     ctypes.c_uint64(cnt), ptr(vb["tau"]))
        assert ok == 1, "first-fill validity <-> settlement link failed"
        o.orc_plonk_pk_free(ctypes.c_void_p(pk_s))
        o.orc_plonk_pk_free(ctypes.c_void_p(pk_f))


@pytest.mark.gpu
class TestBundleLinkGpu:
    """GPU end-to-end production bundle: settlement + both validity proofs on
    the HIP prover, cross-domain links computed on the GPU, verified by the
    oracle's trapdoor verifier."""

    def test_gpu_bundle(self, vb, orc):
        from renegade_amd import load_prover
        plib = load_prover()
        if not plib.gpu_available:
            pytest.skip("no GPU")
        lib = vb["lib"]
        o = vb["o"]
        lib.rng_preprocess.restype = ctypes.c_void_p
        lib.rng_preprocess.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        lib.rng_prove.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                                  ctypes.c_uint64, U64P, U64P]
        lib.rng_link_proofs.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                                        ctypes.c_uint64, ctypes.c_uint64,
                                        ctypes.c_uint64, U64P]
        ts = vb["fetch"](lib.rng_circ_build_settlement_bundle(7))
        t0 = vb["fetch"](lib.rng_circ_build_validity(7, 0))
        t1 = vb["fetch"](lib.rng_circ_build_validity(7, 1))
        u0 = vb["fetch"](lib.rng_circ_build_ob_validity(7, 0))
        u1 = vb["fetch"](lib.rng_circ_build_ob_validity(7, 1))
        n_big = max(int(ts["n"]), int(t0["n"]), int(u0["n"]))
        power = max(4, int(n_big).bit_length())  # covers n_big + 2
        ptau = orc.srs_generate_ptau(power, seed=42)
        max_degree = (1 << power) + 2
        ctx = plib.init(ptau, max_degree)

        class Desc(ctypes.Structure):
            _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                        ("selectors", U64P), ("sigma", U64P),
                        ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

        def gprove(t, seed):
            pk = lib.rng_preprocess(ctx.h, ctypes.byref(
                Desc(t["n"], t["npub"], ptr(t["sel"]), ptr(t["sigma"]), 0, None)))
            assert pk
            proof = np.zeros(157, dtype=np.uint64)
            hint = np.zeros(4 * (int(t["n"]) + 2) + 9, dtype=np.uint64)
            assert lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(t["wires"]),
                                 ptr(t["pubs"]), seed, ptr(proof), ptr(hint)) == 0
            return pk, proof, hint

        pk_s, _, hs = gprove(ts, 3)
        pk_v, _, h0 = gprove(t0, 4)
        _, _, h1 = gprove(t1, 5)
        pk_u, _, g0 = gprove(u0, 6)
        _, _, g1 = gprove(u1, 7)

        def ext(h, n_small):
            out = np.zeros(4 * (n_big + 2) + 9, dtype=np.uint64)
            out[:4 * (n_small + 2)] = h[:4 * (n_small + 2)]
            out[-9:] = h[-9:]
            return out

        hs_e = ext(hs, int(ts["n"]))
        pk_big = {int(ts["n"]): pk_s, int(t0["n"]): pk_v, int(u0["n"]): pk_u}[n_big]
        t_big = {int(ts["n"]): ts, int(t0["n"]): t0, int(u0["n"]): u0}[n_big]
        opk_big = vb["setup"](t_big)
        groups = bundle_groups(ts["lg"])
        legs = [(("party", 0), ext(h0, int(t0["n"]))),
                (("party", 1), ext(h1, int(t1["n"]))),
                (("out", 0), ext(g0, int(u0["n"]))),
                (("out", 1), ext(g1, int(u1["n"])))]
        for key, hv_e in legs:
            a, off, cnt = groups[key]
            lp_g = np.zeros(18, dtype=np.uint64)
            assert lib.rng_link_proofs(ctx.h, ctypes.c_void_p(pk_big), ptr(hv_e),
                                       ptr(hs_e), a, off, cnt, ptr(lp_g)) == 0
            # oracle computes the same link proof bit-exact
            lp_o = np.zeros(18, dtype=np.uint64)
            assert o.orc_plonk_link(ctypes.c_void_p(opk_big), ptr(hv_e), ptr(hs_e),
                                    ctypes.c_uint64(a), ctypes.c_uint64(off),
                                    ctypes.c_uint64(cnt), ptr(lp_o)) == 0
            assert np.array_equal(lp_g, lp_o), f"{key} GPU link != oracle"
            ok = o.orc_plonk_link_verify(ctypes.c_void_p(opk_big),
                                         ptr(hv_e[-9:].copy()), ptr(hs_e[-9:].copy()),
                                         ptr(lp_g), ctypes.c_uint64(a),
                                         ctypes.c_uint64(off), ctypes.c_uint64(cnt),
                                         ptr(vb["tau"]))
            assert ok == 1
        o.orc_plonk_pk_free(ctypes.c_void_p(opk_big))
        ctx.close()
