"""Oracle core parity vs the pure-Python BN254 model (no GPU).

Pins the C++ oracle's field arithmetic, G1 group math, NTT, MSM, keccak256
and SRS generation/parsing against independent bignum computation
(SURVEY.md §8c pinning plan).
"""
import random

import numpy as np
import pytest

from tests import py_ref as ref


def rand_fr(rng):
    return rng.randrange(ref.R)


def rand_fq(rng):
    return rng.randrange(ref.Q)


class TestField:
    def test_fr_mont_roundtrip_and_mul(self, orc):
        rng = random.Random(42)
        for _ in range(200):
            a, b = rand_fr(rng), rand_fr(rng)
            am, bm = orc.fr_from_canonical(a), orc.fr_from_canonical(b)
            assert am == ref.to_mont(a, ref.R)
            assert orc.fr_to_canonical(am) == a
            assert orc.fr_mul(am, bm) == ref.to_mont(a * b % ref.R, ref.R)
            assert orc.fr_add(am, bm) == ref.to_mont((a + b) % ref.R, ref.R)
            assert orc.fr_sub(am, bm) == ref.to_mont((a - b) % ref.R, ref.R)

    def test_fr_inverse(self, orc):
        rng = random.Random(7)
        for _ in range(20):
            a = rand_fr(rng) or 1
            am = orc.fr_from_canonical(a)
            assert orc.fr_inv(am) == ref.to_mont(pow(a, -1, ref.R), ref.R)

    def test_fq_mul_edge_cases(self, orc):
        rng = random.Random(3)
        cases = [0, 1, ref.Q - 1, ref.Q - 2, (1 << 253) % ref.Q] + [rand_fq(rng) for _ in range(100)]
        for a in cases:
            for b in [0, 1, ref.Q - 1, rand_fq(rng)]:
                am, bm = orc.fq_from_canonical(a), orc.fq_from_canonical(b)
                assert orc.fq_mul(am, bm) == ref.to_mont(a * b % ref.Q, ref.Q)

    def test_fq_inverse(self, orc):
        rng = random.Random(9)
        for _ in range(10):
            a = rand_fq(rng) or 1
            am = orc.fq_from_canonical(a)
            assert orc.fq_inv(am) == ref.to_mont(pow(a, -1, ref.Q), ref.Q)


class TestG1:
    def test_generator_on_curve(self, orc):
        g = orc.g1_generator()
        assert orc.g1_is_on_curve(g)
        assert orc.affine_to_py(g) == ref.G1_GEN

    def test_add_vs_ref(self, orc):
        rng = random.Random(11)
        g = ref.G1_GEN
        pts = [ref.ec_mul(g, rng.randrange(1, ref.R)) for _ in range(8)]
        for a in pts[:4]:
            for b in pts[4:]:
                got = orc.affine_to_py(orc.g1_add(orc.affine_from_py(a), orc.affine_from_py(b)))
                assert got == ref.ec_add(a, b)
        # doubling and inverse cases
        p = pts[0]
        assert orc.affine_to_py(orc.g1_add(orc.affine_from_py(p), orc.affine_from_py(p))) == ref.ec_add(p, p)
        neg = (p[0], ref.Q - p[1])
        assert orc.affine_to_py(orc.g1_add(orc.affine_from_py(p), orc.affine_from_py(neg))) is None

    def test_mul_vs_ref(self, orc):
        rng = random.Random(13)
        g = orc.g1_generator()
        for _ in range(6):
            k = rng.randrange(ref.R)
            assert orc.affine_to_py(orc.g1_mul(g, k)) == ref.ec_mul(ref.G1_GEN, k)
        assert orc.affine_to_py(orc.g1_mul(g, 0)) is None
        assert orc.affine_to_py(orc.g1_mul(g, ref.R)) is None  # r*G = identity


class TestNTT:
    @pytest.mark.parametrize("n", [2, 8, 64, 256])
    def test_forward_vs_direct(self, orc, n):
        rng = random.Random(100 + n)
        vals = [rand_fr(rng) for _ in range(n)]
        data = np.zeros(4 * n, dtype=np.uint64)
        for i, v in enumerate(vals):
            m = ref.to_mont(v, ref.R)
            data[4 * i:4 * i + 4] = ref.int_to_limbs(m)
        orc.ntt(data, n)
        expect = ref.ntt_direct(vals)
        for i in range(n):
            got = ref.limbs_to_int(data[4 * i:4 * i + 4])
            assert ref.from_mont(got, ref.R) == expect[i], f"mismatch at {i}"

    @pytest.mark.parametrize("n", [4, 128, 1024])
    def test_roundtrip(self, orc, n):
        rng = random.Random(200 + n)
        data = np.zeros(4 * n, dtype=np.uint64)
        orig = []
        for i in range(n):
            v = rand_fr(rng)
            orig.append(v)
            data[4 * i:4 * i + 4] = ref.int_to_limbs(ref.to_mont(v, ref.R))
        orc.ntt(data, n)
        orc.ntt(data, n, inverse=True)
        for i in range(n):
            got = ref.limbs_to_int(data[4 * i:4 * i + 4])
            assert ref.from_mont(got, ref.R) == orig[i]


    def test_convolution_theorem(self, orc):
        """NTT(a) * NTT(b) pointwise == NTT(a conv b): pins the transform as
        a true DFT over Fr, not just an involution."""
        import random
        n = 64
        rng = random.Random(9)
        a = [rng.randrange(ref.R) for _ in range(n // 2)] + [0] * (n // 2)
        b = [rng.randrange(ref.R) for _ in range(n // 2)] + [0] * (n // 2)
        conv = [0] * n
        for i in range(n // 2):
            for j in range(n // 2):
                conv[i + j] = (conv[i + j] + a[i] * b[j]) % ref.R

        def mont_arr(vals):
            out = np.zeros(4 * n, dtype=np.uint64)
            for i, v in enumerate(vals):
                out[4 * i:4 * i + 4] = ref.int_to_limbs(ref.to_mont(v, ref.R))
            return out

        fa, fb, fc = mont_arr(a), mont_arr(b), mont_arr(conv)
        orc.ntt(fa, n)
        orc.ntt(fb, n)
        orc.ntt(fc, n)
        for i in range(n):
            va = ref.from_mont(ref.limbs_to_int(fa[4 * i:4 * i + 4]), ref.R)
            vb = ref.from_mont(ref.limbs_to_int(fb[4 * i:4 * i + 4]), ref.R)
            vc = ref.from_mont(ref.limbs_to_int(fc[4 * i:4 * i + 4]), ref.R)
            assert va * vb % ref.R == vc, f"slot {i}"

class TestMSM:
    def _mk(self, orc, n, seed):
        rng = random.Random(seed)
        pts, bases, scalars = [], np.zeros(9 * n, dtype=np.uint64), np.zeros(4 * n, dtype=np.uint64)
        ss = []
        for i in range(n):
            k = rng.randrange(1, ref.R)
            p = ref.ec_mul(ref.G1_GEN, k)
            s = rng.randrange(ref.R)
            pts.append(p)
            ss.append(s)
            bases[9 * i:9 * i + 9] = orc.affine_from_py(p)
            scalars[4 * i:4 * i + 4] = ref.int_to_limbs(s)
        return pts, ss, bases, scalars

    @pytest.mark.parametrize("n", [1, 2, 17, 64])
    def test_pippenger_vs_ref(self, orc, n):
        pts, ss, bases, scalars = self._mk(orc, n, 300 + n)
        got = orc.affine_to_py(orc.msm(bases, scalars, n))
        naive = orc.affine_to_py(orc.msm(bases, scalars, n, naive=True))
        expect = ref.msm_ref(pts, ss)
        assert got == expect
        assert naive == expect

    def test_zero_scalars(self, orc):
        pts, ss, bases, scalars = self._mk(orc, 4, 999)
        scalars[:] = 0
        assert orc.affine_to_py(orc.msm(bases, scalars, 4)) is None

    @pytest.mark.parametrize("c", [8, 13, 16])
    def test_window_sizes(self, orc, c):
        pts, ss, bases, scalars = self._mk(orc, 33, 400 + c)
        got = orc.affine_to_py(orc.msm(bases, scalars, 33, window_c=c))
        assert got == ref.msm_ref(pts, ss)


class TestKeccak:
    def test_known_vectors(self, orc):
        # published keccak-256 known-answer vectors (Ethereum variant)
        assert orc.keccak256(b"").hex() == \
            "c5d2460186f7233c927e7db2dcc703c0e500b653ca82273b7bfad8045d85a470"
        assert orc.keccak256(b"abc").hex() == \
            "4e03657aea45a94fc7d47ba826c8d667c0d1e6e33a64a036ec44f58fa12d6c45"
        assert orc.keccak256(b"testing").hex() == \
            "5f16f4c7f149ac4f9510d9cf8cf384038ad348b3bcdc01915f95de12df9d1b02"

    def test_long_input(self, orc):
        # > one rate block (136 B) to exercise multi-block absorb
        data = bytes(range(256)) * 3
        h1 = orc.keccak256(data)
        h2 = orc.keccak256(data)
        assert h1 == h2 and len(h1) == 32
        assert h1 != orc.keccak256(data[:-1])


class TestSRS:
    def test_generate_parse_roundtrip_small(self, orc):
        power = 6  # 64+3 points — keep CPU suite fast
        data = orc.srs_generate_ptau(power, seed=42)
        max_degree = (1 << power) + 2
        g1, h, bh = orc.srs_parse(data, max_degree)
        # point 0 is the generator
        assert orc.affine_to_py(g1[0]) == ref.G1_GEN
        # successive ratio = tau everywhere: check tau from point1 dlog is
        # impossible; instead verify g1[i+1] == tau * g1[i] via python EC with
        # tau recovered from the known derivation
        import hashlib  # noqa: F401  (tau derivation uses keccak, recompute via oracle)
        tau_bytes = orc.keccak256(b"renegade-amd-srs-tau" + (42).to_bytes(8, "little"))
        tau = int.from_bytes(tau_bytes, "little") % ref.R
        for i in [0, 1, 5, max_degree - 1]:
            p = orc.affine_to_py(g1[i])
            q = orc.affine_to_py(g1[i + 1])
            assert q == ref.ec_mul(p, tau)
        # determinism
        assert data == orc.srs_generate_ptau(power, seed=42)
        assert data != orc.srs_generate_ptau(power, seed=43)

    def test_parse_rejects_corruption(self, orc):
        data = bytearray(orc.srs_generate_ptau(5, seed=1))
        data[40] ^= 1  # corrupt inside section 1 modulus area
        with pytest.raises(AssertionError):
            orc.srs_parse(bytes(data), (1 << 5) + 2)
