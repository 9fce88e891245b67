"""ctypes wrapper over librenegade_prover.so (the product C ABI).

Mirrors the reference's prover call surface for this path:
the names and argument meanings follow include/rng_prover.h, which cites the
reference interface each entry point replaces.
"""
import ctypes
import os
from pathlib import Path

import numpy as np

_U64P = ctypes.POINTER(ctypes.c_uint64)
_U8P = ctypes.POINTER(ctypes.c_uint8)

PROOF_U64S = 157


class ProverError(RuntimeError):
    pass


def _lib_path() -> Path:
    here = Path(__file__).resolve().parent
    return here / "librenegade_prover.so"


def load_prover():
    """Load the HIP prover library; raises loudly if absent or unbuilt."""
    p = _lib_path()
    if not p.exists():
        raise ProverError(
            f"HIP prover extension not found at {p}. Build it with "
            f"`python -c 'import __graft_entry__; __graft_entry__.build()'` — "
            f"there is no CPU fallback."
        )
    return ProverLib(str(p))


def _ptr(a):
    return a.ctypes.data_as(_U64P)


class ProverLib:
    def __init__(self, path):
        self.lib = ctypes.CDLL(path)
        self.lib.rng_prover_init.restype = ctypes.c_void_p
        self.lib.rng_prover_init.argtypes = [_U8P, ctypes.c_size_t, ctypes.c_uint64]
        self.lib.rng_ctx_free.argtypes = [ctypes.c_void_p]
        self.lib.rng_version.restype = ctypes.c_char_p
        self.lib.rng_dbuf_alloc.restype = ctypes.c_void_p
        self.lib.rng_dbuf_alloc.argtypes = [ctypes.c_size_t]
        self.lib.rng_dbuf_free.argtypes = [ctypes.c_void_p]
        self.lib.rng_dbuf_upload.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
        self.lib.rng_dbuf_download.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
        self.lib.rng_ntt_fr.argtypes = [ctypes.c_void_p, _U64P, ctypes.c_uint64,
                                        ctypes.c_uint64, ctypes.c_int]
        self.lib.rng_ntt_fr_dev.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                            ctypes.c_uint64, ctypes.c_uint64, ctypes.c_int]
        self.lib.rng_ntt_fr_dev_oop.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                                ctypes.c_void_p, ctypes.c_uint64,
                                                ctypes.c_uint64, ctypes.c_int]
        self.lib.rng_msm_g1.argtypes = [ctypes.c_void_p, _U64P, _U64P, ctypes.c_uint64,
                                        _U64P, ctypes.c_int]
        self.lib.rng_msm_g1_dev.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                            ctypes.c_void_p, ctypes.c_uint64, _U64P,
                                            ctypes.c_int]
        self.lib.rng_srs_dev_bases.restype = ctypes.c_void_p
        self.lib.rng_srs_dev_bases.argtypes = [ctypes.c_void_p, _U64P]

    @property
    def version(self) -> str:
        return self.lib.rng_version().decode()

    def set_device(self, dev: int):
        rc = self.lib.rng_set_device(dev)
        if rc != 0:
            raise ProverError(f"rng_set_device({dev}) failed rc={rc}")

    def msm_last_times(self):
        out = (ctypes.c_double * 5)()
        self.lib.rng_msm_last_times(out)
        return {"digits": out[0], "sort": out[1], "bucket_reduce": out[2],
                "window_chunks": out[3], "final": out[4]}

    def ntt_last_times(self):
        out = (ctypes.c_double * 2)()
        self.lib.rng_ntt_last_times(out)
        return {"pass1": out[0], "pass2": out[1]}

    @property
    def gpu_available(self) -> bool:
        return bool(self.lib.rng_gpu_available())

    def require_gpu(self):
        if not self.gpu_available:
            raise ProverError("no AMD GPU visible — the product path requires MI355X")

    # ---- context ----
    def init(self, srs_ptau: bytes, max_degree: int) -> "ProverCtx":
        buf = (ctypes.c_uint8 * len(srs_ptau)).from_buffer_copy(srs_ptau)
        h = self.lib.rng_prover_init(buf, len(srs_ptau), max_degree)
        if not h:
            raise ProverError("rng_prover_init failed (bad SRS?)")
        return ProverCtx(self, h)


class ProverCtx:
    def __init__(self, plib: ProverLib, handle):
        self._plib = plib
        self.lib = plib.lib
        self.h = handle

    def close(self):
        if self.h:
            self.lib.rng_ctx_free(self.h)
            self.h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def _check(self, rc, what):
        if rc != 0:
            raise ProverError(f"{what} failed rc={rc}")

    # ---- primitives ----
    def ntt(self, data: np.ndarray, n: int, batch: int = 1, inverse: bool = False):
        """In-place NTT on a host uint64 array of 4*n*batch limbs."""
        assert data.dtype == np.uint64 and data.size == 4 * n * batch
        self._check(self.lib.rng_ntt_fr(self.h, _ptr(data), n, batch,
                                        1 if inverse else 0), "rng_ntt_fr")

    def msm(self, bases: np.ndarray, scalars: np.ndarray, n: int, window_c: int = 0):
        """bases: 8*n u64 (x,y Montgomery); scalars: 4*n u64 canonical.
        Returns 9-u64 affine record."""
        assert bases.dtype == np.uint64 and bases.size == 8 * n
        assert scalars.dtype == np.uint64 and scalars.size == 4 * n
        out = np.zeros(9, dtype=np.uint64)
        self._check(self.lib.rng_msm_g1(self.h, _ptr(bases), _ptr(scalars), n, _ptr(out),
                                        window_c), "rng_msm_g1")
        return out

    # ---- device-resident helpers (benchmarking) ----
    def dbuf_from(self, host: np.ndarray):
        nbytes = host.nbytes
        d = self.lib.rng_dbuf_alloc(nbytes)
        if not d:
            raise ProverError("rng_dbuf_alloc failed")
        self._check(self.lib.rng_dbuf_upload(d, host.ctypes.data_as(ctypes.c_void_p),
                                             nbytes), "upload")
        return d

    def dbuf_alloc(self, nbytes: int):
        d = self.lib.rng_dbuf_alloc(nbytes)
        if not d:
            raise ProverError("rng_dbuf_alloc failed")
        return d

    def dbuf_download(self, d, host: np.ndarray):
        self._check(self.lib.rng_dbuf_download(d, host.ctypes.data_as(ctypes.c_void_p),
                                               host.nbytes), "download")

    def dbuf_free(self, d):
        self.lib.rng_dbuf_free(d)

    def ntt_dev(self, d, n, batch=1, inverse=False):
        self._check(self.lib.rng_ntt_fr_dev(self.h, d, n, batch, 1 if inverse else 0),
                    "rng_ntt_fr_dev")

    def ntt_dev_oop(self, din, dout, n, batch=1, inverse=False):
        self._check(self.lib.rng_ntt_fr_dev_oop(self.h, din, dout, n, batch,
                                                1 if inverse else 0), "rng_ntt_fr_dev_oop")

    def msm_dev(self, dbases, dscalars, n, window_c=0):
        out = np.zeros(9, dtype=np.uint64)
        self._check(self.lib.rng_msm_g1_dev(self.h, dbases, dscalars, n, _ptr(out),
                                            window_c), "rng_msm_g1_dev")
        return out

    def sync(self):
        self._check(self.lib.rng_device_sync(), "rng_device_sync")
