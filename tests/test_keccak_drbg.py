"""Keccak-256 + blinder DRBG: the PRODUCT library's implementations pinned
directly against published keccak vectors and against the ORACLE's
independent implementation (both were bitten once by the rol(x,0) UB — this
test guards the exact seam)."""
import ctypes

import numpy as np
import pytest

U8P = ctypes.POINTER(ctypes.c_uint8)
U64P = ctypes.POINTER(ctypes.c_uint64)

# published Keccak-256 (NOT SHA3-256) known-answer vectors
VECTORS = [
    (b"", "c5d2460186f7233c927e7db2dcc703c0e500b653ca82273b7bfad8045d85a470"),
    (b"abc", "4e03657aea45a94fc7d47ba826c8d667c0d1e6e33a64a036ec44f58fa12d6c45"),
    (b"testing", "5f16f4c7f149ac4f9510d9cf8cf384038ad348b3bcdc01915f95de12df9d1b02"),
]


@pytest.fixture(scope="module")
def libs(orc):
    from renegade_amd import load_prover
    plib = load_prover()
    lib = plib.lib
    lib.rng_keccak256.argtypes = [U8P, ctypes.c_size_t, U8P]
    orc.lib.orc_keccak256.argtypes = [U8P, ctypes.c_uint64, U8P]
    lib.rng_debug_drbg.argtypes = [ctypes.c_uint64, ctypes.c_uint32, U64P]
    return lib, orc.lib


def _hash(fn, data):
    buf = (ctypes.c_uint8 * max(1, len(data)))(*data)
    out = (ctypes.c_uint8 * 32)()
    fn(buf, len(data), out)
    return bytes(out).hex()


def test_product_keccak_known_vectors(libs):
    lib, _ = libs
    for msg, want in VECTORS:
        assert _hash(lib.rng_keccak256, msg) == want, msg


def test_oracle_keccak_known_vectors(libs):
    _, o = libs
    for msg, want in VECTORS:
        assert _hash(o.orc_keccak256, msg) == want, msg


def test_product_vs_oracle_random(libs):
    lib, o = libs
    rng = np.random.default_rng(3)
    for ln in [1, 31, 32, 33, 135, 136, 137, 1000]:
        data = bytes(rng.integers(0, 256, ln, dtype=np.uint8))
        assert _hash(lib.rng_keccak256, data) == _hash(o.orc_keccak256, data)


def test_drbg_spec(libs):
    """Blinder DRBG spec (oracle/transcript.hpp, normative): block i =
    LE(keccak256("rng-blind" || le64(seed) || le32(i))) mod r."""
    lib, _ = libs
    import hashlib  # no keccak in hashlib; recompute via the product hash
    seed, nblocks = 42, 4
    out = np.zeros(4 * nblocks, dtype=np.uint64)
    lib.rng_debug_drbg(seed, nblocks, out.ctypes.data_as(U64P))
    R = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    for i in range(nblocks):
        msg = b"rng-blind" + seed.to_bytes(8, "little") + i.to_bytes(4, "little")
        h = bytes.fromhex(_hash(lib.rng_keccak256, msg))
        want = int.from_bytes(h, "little") % R
        # the export emits Fr values in MONTGOMERY form (library convention)
        got_mont = sum(int(out[4 * i + k]) << (64 * k) for k in range(4))
        got = got_mont * pow(1 << 256, -1, R) % R
        assert got == want, f"block {i}"
