"""GPU parity tests: HIP MSM/NTT kernels vs the CPU oracle, bit-exact.

These are the parity tests proper (SURVEY.md §8c contract (i)): same inputs,
byte-identical outputs in the pinned 4xu64-LE-Montgomery layout.
"""
import random

import numpy as np
import pytest

from tests import py_ref as ref

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def plib():
    from renegade_amd import load_prover
    lib = load_prover()
    if not lib.gpu_available:
        pytest.skip("no GPU visible")
    return lib


@pytest.fixture(scope="module")
def srs10(orc):
    power = 10
    ptau = orc.srs_generate_ptau(power, seed=42)
    g1, h, bh = orc.srs_parse(ptau, (1 << power) + 2)
    return ptau, g1


@pytest.fixture(scope="module")
def ctx(plib, srs10):
    ptau, _ = srs10
    c = plib.init(ptau, (1 << 10) + 2)
    yield c
    c.close()


def mk_scalars(n, seed, special=None):
    rng = random.Random(seed)
    out = np.zeros(4 * n, dtype=np.uint64)
    vals = []
    for i in range(n):
        if special is not None and i < len(special):
            s = special[i]
        else:
            s = rng.randrange(ref.R)
        vals.append(s)
        out[4 * i:4 * i + 4] = ref.int_to_limbs(s)
    return out, vals


def mk_fr_mont(n, seed):
    rng = random.Random(seed)
    data = np.zeros(4 * n, dtype=np.uint64)
    for i in range(n):
        data[4 * i:4 * i + 4] = ref.int_to_limbs(ref.to_mont(rng.randrange(ref.R), ref.R))
    return data


class TestMsmGpu:
    def _bases(self, orc, g1, n, tile=True):
        """Take n bases from the SRS (repeat/tile if needed)."""
        have = g1.shape[0]
        idx = [i % have for i in range(n)]
        b9 = np.ascontiguousarray(g1[idx]).reshape(-1)
        b8 = np.ascontiguousarray(g1[idx][:, :8]).reshape(-1)
        return b8, b9

    # 70000 exercises the c=13 auto-window tier (2^16 < n <= 2^18), which no
    # proof workload hits
    @pytest.mark.parametrize("n", [1, 2, 255, 4096, 70000])
    def test_parity_sizes(self, ctx, orc, srs10, n):
        _, g1 = srs10
        b8, b9 = self._bases(orc, g1, n)
        scalars, _ = mk_scalars(n, 1000 + n)
        got = ctx.msm(b8, scalars, n)
        expect = orc.msm(b9, scalars, n)
        assert np.array_equal(got, expect)

    def test_edge_scalars(self, ctx, orc, srs10):
        _, g1 = srs10
        n = 64
        b8, b9 = self._bases(orc, g1, n)
        special = [0, 1, ref.R - 1, (1 << 253) - 1, 0xFFFF, 0x8000, 0x18000,
                   (1 << 128) - 1, ref.R - 2, 2]
        scalars, _ = mk_scalars(n, 5, special=special)
        got = ctx.msm(b8, scalars, n)
        expect = orc.msm(b9, scalars, n)
        assert np.array_equal(got, expect)

    def test_all_zero(self, ctx, orc, srs10):
        _, g1 = srs10
        n = 32
        b8, b9 = self._bases(orc, g1, n)
        scalars = np.zeros(4 * n, dtype=np.uint64)
        got = ctx.msm(b8, scalars, n)
        assert got[8] == 1  # identity

    def test_duplicate_scalars_bucket_collisions(self, ctx, orc, srs10):
        _, g1 = srs10
        n = 512
        b8, b9 = self._bases(orc, g1, n)
        rng = random.Random(77)
        # few distinct scalar values -> heavy bucket collisions
        vals = [rng.randrange(ref.R) for _ in range(4)]
        special = [vals[i % 4] for i in range(n)]
        scalars, _ = mk_scalars(n, 0, special=special)
        got = ctx.msm(b8, scalars, n)
        expect = orc.msm(b9, scalars, n)
        assert np.array_equal(got, expect)

    @pytest.mark.parametrize("c", [8, 12, 16])
    def test_window_sizes(self, ctx, orc, srs10, c):
        _, g1 = srs10
        n = 300
        b8, b9 = self._bases(orc, g1, n)
        scalars, _ = mk_scalars(n, 2000 + c)
        got = ctx.msm(b8, scalars, n, window_c=c)
        expect = orc.msm(b9, scalars, n)
        assert np.array_equal(got, expect)


class TestNttGpu:
    @pytest.mark.parametrize("logn", [2, 6, 12, 13, 14, 16, 18])
    def test_forward_parity(self, ctx, orc, logn):
        n = 1 << logn
        data = mk_fr_mont(n, 3000 + logn)
        data_orc = data.copy()
        ctx.ntt(data, n)
        orc.ntt(data_orc, n)
        assert np.array_equal(data, data_orc)

    @pytest.mark.parametrize("logn", [12, 17])
    def test_inverse_parity(self, ctx, orc, logn):
        n = 1 << logn
        data = mk_fr_mont(n, 4000 + logn)
        data_orc = data.copy()
        ctx.ntt(data, n, inverse=True)
        orc.ntt(data_orc, n, inverse=True)
        assert np.array_equal(data, data_orc)

    def test_roundtrip_2_22(self, ctx):
        n = 1 << 22
        rng = np.random.default_rng(9)
        # build Montgomery-valid elements from random 253-bit ints (below r)
        data = rng.integers(0, 1 << 63, size=4 * n, dtype=np.uint64)
        data[3::4] &= (1 << 61) - 1  # force < 2^253 < r... top limb < 2^61
        orig = data.copy()
        ctx.ntt(data, n)
        assert not np.array_equal(data, orig)
        ctx.ntt(data, n, inverse=True)
        assert np.array_equal(data, orig)

    def test_batch(self, ctx, orc):
        n, batch = 1 << 12, 3
        data = mk_fr_mont(n * batch, 5000)
        data_orc = data.copy()
        ctx.ntt(data, n, batch=batch)
        for b in range(batch):
            orc.ntt(data_orc[4 * n * b:4 * n * (b + 1)], n)
        assert np.array_equal(data, data_orc)

    def test_linearity(self, ctx):
        # NTT(a) + NTT(b) == NTT(a+b) — size-independent property at full size
        n = 1 << 20
        rng = np.random.default_rng(11)
        # all limbs < 2^61 => values < 2^253 < r, and a+b has no per-limb
        # carry and stays < r: limb-wise uint64 add IS the field add
        a = rng.integers(0, 1 << 61, size=4 * n, dtype=np.uint64)
        b = rng.integers(0, 1 << 61, size=4 * n, dtype=np.uint64)
        s = a + b
        ctx.ntt(a, n)
        ctx.ntt(b, n)
        ctx.ntt(s, n)
        for i in [0, 1, n // 2, n - 1]:
            va = ref.limbs_to_int(a[4 * i:4 * i + 4])
            vb = ref.limbs_to_int(b[4 * i:4 * i + 4])
            vs = ref.limbs_to_int(s[4 * i:4 * i + 4])
            assert (va + vb) % ref.R == vs
