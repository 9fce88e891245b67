"""State-rotation circuits (`Valid Deposit`, `Valid Withdrawal`,
`Valid Order Cancellation` — zk_circuits/valid_deposit.rs,
valid_withdrawal.rs, valid_order_cancellation.rs): build, satisfiability,
oracle prove/verify, tamper rejection (CPU)."""
import ctypes

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

# builder -> (statement scalars, tampered public-input word offsets)
BUILDER_INFO = {
    "rng_circ_build_valid_deposit": (8, [3 * 4, 4 * 4, 5 * 4, 6 * 4]),
    "rng_circ_build_valid_withdrawal": (8, [3 * 4, 4 * 4, 5 * 4, 6 * 4]),
    "rng_circ_build_valid_order_cancellation": (3, [0, 1 * 4, 2 * 4]),
    "rng_circ_build_note_redemption": (6, [4 * 4, 5 * 4]),
    "rng_circ_build_fee_public_relayer": (9, [0, 1 * 4, 2 * 4, 3 * 4]),
    "rng_circ_build_fee_public_protocol": (9, [0, 1 * 4, 2 * 4, 3 * 4]),
    "rng_circ_build_fee_private_relayer": (7, [0, 1 * 4, 2 * 4, 3 * 4, 6 * 4]),
}
BUILDERS = list(BUILDER_INFO)


@pytest.fixture(scope="module")
def plib():
    from renegade_amd import load_prover
    return load_prover()


def build_tables(lib, builder, seed):
    fn = getattr(lib, builder)
    fn.restype = ctypes.c_void_p
    fn.argtypes = [ctypes.c_uint64]
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]
    h = fn(seed)
    assert h, f"{builder} unsatisfied"
    n = lib.rng_circ_n(h)
    npub = lib.rng_circ_npub(h)
    sel = np.zeros(13 * n * 4, dtype=np.uint64)
    sigma = np.zeros(5 * n, dtype=np.uint64)
    wires = np.zeros(5 * n * 4, dtype=np.uint64)
    pubs = np.zeros(npub * 4, dtype=np.uint64)
    lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
    lib.rng_circ_free(h)
    return n, npub, sel, sigma, wires, pubs


@pytest.mark.parametrize("builder", BUILDERS)
def test_rotation_circuit_prove_verify(plib, orc, builder):
    n, npub, sel, sigma, wires, pubs = build_tables(plib.lib, builder, 42)
    assert npub == BUILDER_INFO[builder][0]
    power = max(4, int(n).bit_length())
    ptau = orc.srs_generate_ptau(power, seed=42)
    md = (1 << power) + 2
    g1, _, _ = orc.srs_parse(ptau, md)
    srs = np.ascontiguousarray(g1).reshape(-1)
    o = orc.lib
    o.orc_plonk_preprocess.restype = ctypes.c_void_p
    o.orc_plonk_preprocess.argtypes = [ctypes.c_uint64, ctypes.c_uint64, U64P, U64P,
                                       U64P, ctypes.c_uint64]
    o.orc_plonk_prove.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P]
    o.orc_plonk_verify.argtypes = [ctypes.c_void_p, U64P, U64P, U64P]
    o.orc_derive_tau.argtypes = [ctypes.c_uint64, U64P]
    pk = o.orc_plonk_preprocess(n, npub, ptr(sel), ptr(sigma), ptr(srs), md + 1)
    assert pk
    proof = np.zeros(157, dtype=np.uint64)
    assert o.orc_plonk_prove(ctypes.c_void_p(pk), ptr(wires), ptr(pubs), 7,
                             ptr(proof)) == 0
    tau = np.zeros(4, dtype=np.uint64)
    o.orc_derive_tau(42, ptr(tau))
    assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(pubs), ptr(proof), ptr(tau)) == 1
    for idx in BUILDER_INFO[builder][1]:
        bad = pubs.copy()
        bad[idx] ^= np.uint64(1)
        assert o.orc_plonk_verify(ctypes.c_void_p(pk), ptr(bad), ptr(proof),
                                  ptr(tau)) != 1
    o.orc_plonk_pk_free(ctypes.c_void_p(pk))


@pytest.mark.parametrize("builder", BUILDERS)
def test_seeds_vary(plib, builder):
    for seed in [1, 9]:
        fn = getattr(plib.lib, builder)
        fn.restype = ctypes.c_void_p
        fn.argtypes = [ctypes.c_uint64]
        h = fn(seed)
        assert h
        plib.lib.rng_circ_free(h)
