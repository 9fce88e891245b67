"""The generic from-scalars arithmetization (`rng_circ_from_scalars`, the
prover-service request shape): every kind round-trips its fixed-seed test
vector into a satisfied circuit, a tampered statement is rejected, and the
rebuilt tables match the native builder's where one exists (CPU only)."""
import ctypes

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

ALL_KINDS = [1, 2, 3, 4, 5, 6, 7, 8, 9, 11, 12, 13, 14, 15, 16, 17, 18, 19]


@pytest.fixture(scope="module")
def fs():
    from renegade_amd import load_prover
    lib = load_prover().lib
    lib.rng_ws_sizes.restype = ctypes.c_int
    lib.rng_ws_sizes.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_uint64),
                                 ctypes.POINTER(ctypes.c_uint64)]
    lib.rng_witness_statement.restype = ctypes.c_int
    lib.rng_witness_statement.argtypes = [ctypes.c_int, ctypes.c_uint64, U64P, U64P]
    lib.rng_circ_from_scalars.restype = ctypes.c_void_p
    lib.rng_circ_from_scalars.argtypes = [ctypes.c_int, U64P, U64P]
    lib.rng_circ_n.restype = ctypes.c_uint64
    lib.rng_circ_n.argtypes = [ctypes.c_void_p]
    lib.rng_circ_npub.restype = ctypes.c_uint64
    lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
    lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
    lib.rng_circ_free.argtypes = [ctypes.c_void_p]
    return lib


def vectors(lib, kind, seed=11):
    nw, ns = ctypes.c_uint64(), ctypes.c_uint64()
    assert lib.rng_ws_sizes(kind, ctypes.byref(nw), ctypes.byref(ns)) == 0
    w = np.zeros(4 * nw.value, dtype=np.uint64)
    s = np.zeros(4 * ns.value, dtype=np.uint64)
    assert lib.rng_witness_statement(kind, seed, ptr(w), ptr(s)) == 0
    return w, s


@pytest.mark.parametrize("kind", ALL_KINDS)
def test_roundtrip_and_tamper(fs, kind):
    lib = fs
    w, s = vectors(lib, kind)
    h = lib.rng_circ_from_scalars(kind, ptr(w), ptr(s))
    assert h, f"kind {kind}: valid vector unsatisfied"
    assert lib.rng_circ_n(h) > 0
    lib.rng_circ_free(h)
    bad = s.copy()
    bad[0] ^= np.uint64(1)
    h2 = lib.rng_circ_from_scalars(kind, ptr(w), ptr(bad))
    assert not h2, f"kind {kind}: tampered statement accepted"


def test_tables_match_native_builder(fs):
    """kind 4 (intent-and-balance validity) must produce byte-identical
    tables to rng_circ_build_validity for the same underlying witness."""
    lib = fs
    lib.rng_circ_build_validity.restype = ctypes.c_void_p
    lib.rng_circ_build_validity.argtypes = [ctypes.c_uint64, ctypes.c_uint64]

    def tables(h):
        assert h
        n = lib.rng_circ_n(h)
        npub = lib.rng_circ_npub(h)
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sigma = np.zeros(5 * n, dtype=np.uint64)
        wires = np.zeros(5 * n * 4, dtype=np.uint64)
        pubs = np.zeros(npub * 4, dtype=np.uint64)
        lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
        lib.rng_circ_free(h)
        return sel, sigma, wires, pubs

    w, s = vectors(lib, 4, seed=9)
    a = tables(lib.rng_circ_from_scalars(4, ptr(w), ptr(s)))
    b = tables(lib.rng_circ_build_validity(9, 0))
    for x, y, what in zip(a, b, ["selectors", "sigma", "wires", "pubs"]):
        assert np.array_equal(x, y), f"{what} differ between builders"


def test_witness_statement_respects_sizes(fs):
    """rng_witness_statement must write exactly rng_ws_sizes scalars (a
    mismatch would silently corrupt the caller's buffers): canary words
    beyond the declared sizes stay untouched."""
    lib = fs
    CANARY = np.uint64(0xDEADBEEFCAFEBABE)
    for kind in ALL_KINDS:
        nw, ns = ctypes.c_uint64(), ctypes.c_uint64()
        assert lib.rng_ws_sizes(kind, ctypes.byref(nw), ctypes.byref(ns)) == 0
        w = np.full(4 * nw.value + 8, CANARY, dtype=np.uint64)
        s = np.full(4 * ns.value + 8, CANARY, dtype=np.uint64)
        assert lib.rng_witness_statement(kind, 3, ptr(w), ptr(s)) == 0
        assert (w[4 * nw.value:] == CANARY).all(), f"kind {kind} overwrote witness"
        assert (s[4 * ns.value:] == CANARY).all(), f"kind {kind} overwrote statement"
        # and the declared region was fully written (no canary residue --
        # scalar values equal to the canary are astronomically unlikely)
        assert (w[:4 * nw.value] != CANARY).any() or nw.value == 0
        assert not (s[:4 * ns.value] == CANARY).all()
