"""ctypes bindings for the CPU oracle (test infrastructure only)."""
import ctypes

import numpy as np

U64P = ctypes.POINTER(ctypes.c_uint64)
U8P = ctypes.POINTER(ctypes.c_uint8)


def arr(n):
    return np.zeros(n, dtype=np.uint64)


def ptr(a):
    return a.ctypes.data_as(U64P)


def bptr(b):
    return (ctypes.c_uint8 * len(b)).from_buffer_copy(b)


class OracleLib:
    def __init__(self, path):
        self.lib = ctypes.CDLL(path)
        self.lib.orc_srs_ptau_size.restype = ctypes.c_uint64
        self.lib.orc_srs_generate_ptau.restype = ctypes.c_uint64
        self.lib.orc_num_threads.restype = ctypes.c_int

    # field helpers operating on python ints ------------------------------
    def _binop(self, fn, a, b):
        aa, bb, oo = arr(4), arr(4), arr(4)
        aa[:] = [(a >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(4)]
        bb[:] = [(b >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(4)]
        fn(ptr(aa), ptr(bb), ptr(oo))
        return int(oo[0]) | int(oo[1]) << 64 | int(oo[2]) << 128 | int(oo[3]) << 192

    def _unop(self, fn, a):
        aa, oo = arr(4), arr(4)
        aa[:] = [(a >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(4)]
        fn(ptr(aa), ptr(oo))
        return int(oo[0]) | int(oo[1]) << 64 | int(oo[2]) << 128 | int(oo[3]) << 192

    def fr_mul(self, a, b):
        return self._binop(self.lib.orc_fr_mul, a, b)

    def fr_add(self, a, b):
        return self._binop(self.lib.orc_fr_add, a, b)

    def fr_sub(self, a, b):
        return self._binop(self.lib.orc_fr_sub, a, b)

    def fr_inv(self, a):
        return self._unop(self.lib.orc_fr_inv, a)

    def fr_from_canonical(self, a):
        return self._unop(self.lib.orc_fr_from_canonical, a)

    def fr_to_canonical(self, a):
        return self._unop(self.lib.orc_fr_to_canonical, a)

    def fq_mul(self, a, b):
        return self._binop(self.lib.orc_fq_mul, a, b)

    def fq_inv(self, a):
        return self._unop(self.lib.orc_fq_inv, a)

    def fq_from_canonical(self, a):
        return self._unop(self.lib.orc_fq_from_canonical, a)

    def fq_to_canonical(self, a):
        return self._unop(self.lib.orc_fq_to_canonical, a)

    # G1 -------------------------------------------------------------------
    def g1_generator(self):
        o = arr(9)
        self.lib.orc_g1_generator(ptr(o))
        return o

    def g1_add(self, a, b):
        o = arr(9)
        self.lib.orc_g1_add(ptr(a), ptr(b), ptr(o))
        return o

    def g1_mul(self, p, scalar_int):
        s = arr(4)
        s[:] = [(scalar_int >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(4)]
        o = arr(9)
        self.lib.orc_g1_mul(ptr(p), ptr(s), ptr(o))
        return o

    def g1_is_on_curve(self, p):
        return bool(self.lib.orc_g1_is_on_curve(ptr(p)))

    def affine_to_py(self, rec):
        """9-u64 affine record (Montgomery) -> python (x, y) canonical or None."""
        if rec[8] != 0:
            return None
        x = self.fq_to_canonical(int(rec[0]) | int(rec[1]) << 64 | int(rec[2]) << 128 | int(rec[3]) << 192)
        y = self.fq_to_canonical(int(rec[4]) | int(rec[5]) << 64 | int(rec[6]) << 128 | int(rec[7]) << 192)
        return (x, y)

    def affine_from_py(self, p):
        rec = arr(9)
        if p is None:
            rec[8] = 1
            return rec
        x = self.fq_from_canonical(p[0])
        y = self.fq_from_canonical(p[1])
        rec[:4] = [(x >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(4)]
        rec[4:8] = [(y >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(4)]
        return rec

    # NTT ------------------------------------------------------------------
    def ntt(self, data_u64, n, inverse=False):
        self.lib.orc_ntt(ptr(data_u64), ctypes.c_uint64(n), ctypes.c_int(1 if inverse else 0))

    # MSM ------------------------------------------------------------------
    def msm(self, bases_u64, scalars_u64, n, window_c=13, naive=False):
        o = arr(9)
        if naive:
            self.lib.orc_msm_naive(ptr(bases_u64), ptr(scalars_u64), ctypes.c_uint64(n), ptr(o))
        else:
            self.lib.orc_msm(ptr(bases_u64), ptr(scalars_u64), ctypes.c_uint64(n), ptr(o),
                             ctypes.c_int(window_c))
        return o

    # keccak ---------------------------------------------------------------
    def keccak256(self, data: bytes) -> bytes:
        buf = bptr(data) if data else (ctypes.c_uint8 * 1)()
        out = (ctypes.c_uint8 * 32)()
        self.lib.orc_keccak256(buf, ctypes.c_uint64(len(data)), out)
        return bytes(out)

    # SRS ------------------------------------------------------------------
    # SRS generation/parsing are pure functions of (power, seed) — memoize
    # per process (many test modules regenerate the same SRS; under a loaded
    # container the repeats dominated suite time)
    _srs_gen_cache = {}
    _srs_parse_cache = {}

    def srs_generate_ptau(self, power, seed) -> bytes:
        key = (power, seed)
        hit = OracleLib._srs_gen_cache.get(key)
        if hit is not None:
            return hit
        size = self.lib.orc_srs_ptau_size(ctypes.c_int(power))
        buf = (ctypes.c_uint8 * size)()
        n = self.lib.orc_srs_generate_ptau(ctypes.c_int(power), ctypes.c_uint64(seed), buf,
                                           ctypes.c_uint64(size))
        assert n != 0, "srs buffer too small"
        out = bytes(buf[:n])
        OracleLib._srs_gen_cache[key] = out
        return out

    def srs_parse(self, data: bytes, max_degree):
        key = (id(data), len(data), max_degree)
        hit = OracleLib._srs_parse_cache.get(key)
        if hit is not None:
            return hit
        g1 = arr(9 * (max_degree + 1))
        h = arr(16)
        bh = arr(16)
        buf = bptr(data)
        rc = self.lib.orc_srs_parse(buf, ctypes.c_uint64(len(data)),
                                    ctypes.c_uint64(max_degree), ptr(g1), ptr(h), ptr(bh))
        assert rc == 0, f"srs parse failed rc={rc}"
        out = (g1.reshape(max_degree + 1, 9), h, bh)
        # key by object identity of the (cached, immutable) bytes + length:
        # safe because generate_ptau returns the same interned object per key
        OracleLib._srs_parse_cache[key] = out
        return out
