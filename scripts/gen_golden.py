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

    # --- link-proof fixture: cross-domain validity(party0) <-> settlement
    #     link of the seed-7 bundle (pins the link layer's bytes: hints,
    #     grid placement, link transcript, quotient + opening) ---
    lib.rng_circ_build_settlement_bundle.restype = ctypes.c_void_p
    lib.rng_circ_build_settlement_bundle.argtypes = [ctypes.c_uint64]
    lib.rng_circ_num_link_groups.restype = ctypes.c_uint64
    lib.rng_circ_num_link_groups.argtypes = [ctypes.c_void_p]
    lib.rng_circ_link_groups.argtypes = [ctypes.c_void_p, U64P]

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
        return dict(n=n, npub=npub, lg=lg.reshape(-1, 3), sel=sel, sigma=sigma,
                    wires=wires, pubs=pubs)

    o.orc_plonk_prove_with_hint.argtypes = [ctypes.c_void_p, U64P, U64P,
                                            ctypes.c_uint64, U64P, U64P]
    o.orc_plonk_link.argtypes = [ctypes.c_void_p, U64P, U64P] + \
        [ctypes.c_uint64] * 3 + [U64P]
    ts = fetch(lib.rng_circ_build_settlement_bundle(7))
    lib.rng_circ_build_validity.restype = ctypes.c_void_p
    lib.rng_circ_build_validity.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
    tv = fetch(lib.rng_circ_build_validity(7, 0))
    n_big = max(int(ts["n"]), int(tv["n"]))
    power = max(4, int(n_big).bit_length())
    ptau = orc.srs_generate_ptau(power, seed=42)
    md = (1 << power) + 2
    g1, _, _ = orc.srs_parse(ptau, md)
    srs = np.ascontiguousarray(g1).reshape(-1)

    def setup_prove(t, bseed):
        pk = o.orc_plonk_preprocess(t["n"], t["npub"], ptr(t["sel"]), ptr(t["sigma"]),
                                    ptr(srs), md + 1)
        assert pk
        proof = np.zeros(157, dtype=np.uint64)
        hint = np.zeros(4 * (int(t["n"]) + 2) + 9, dtype=np.uint64)
        assert o.orc_plonk_prove_with_hint(ctypes.c_void_p(pk), ptr(t["wires"]),
                                           ptr(t["pubs"]), ctypes.c_uint64(bseed),
                                           ptr(proof), ptr(hint)) == 0
        return pk, proof, hint

    pk_s, _, hs = setup_prove(ts, 3)
    pk_v, _, hv = setup_prove(tv, 4)

    def ext(h, n_small):
        out = np.zeros(4 * (n_big + 2) + 9, dtype=np.uint64)
        out[:4 * (n_small + 2)] = h[:4 * (n_small + 2)]
        out[-9:] = h[-9:]
        return out

    pk_big = pk_v if int(tv["n"]) == n_big else pk_s
    groups = sorted((int(r[1]), int(r[0]), int(r[2])) for r in ts["lg"]
                    if int(r[2]) == 17)
    off, align, cnt = groups[0]  # party0
    lp = np.zeros(18, dtype=np.uint64)
    assert o.orc_plonk_link(ctypes.c_void_p(pk_big), ptr(ext(hv, int(tv["n"]))),
                            ptr(ext(hs, int(ts["n"]))), ctypes.c_uint64(align),
                            ctypes.c_uint64(off), ctypes.c_uint64(cnt), ptr(lp)) == 0
    np.save(out_dir / "bundle7_party0_link_proof.npy", lp)
    print(f"link fixture: align={align} off={off} cnt={cnt} lp[0]={lp[0]:#x}")


if __name__ == "__main__":
    main()
