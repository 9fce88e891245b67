"""The PRODUCT path must fail loudly without a GPU (tier contract: no silent
CPU fallback; oracle/ is test infrastructure only).  These run in the
no-GPU container and pin the refusal behavior."""
import ctypes

import numpy as np
import pytest

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

RNG_ERR_NO_GPU = -1


@pytest.fixture(scope="module")
def plib():
    from renegade_amd import load_prover
    return load_prover()


def test_compute_entry_points_refuse_without_gpu(plib):
    if plib.gpu_available:
        pytest.skip("GPU present; refusal paths not reachable")
    lib = plib.lib
    # init refuses outright (SRS upload needs the device)
    lib.rng_prover_init.restype = ctypes.c_void_p
    lib.rng_prover_init.argtypes = [ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
                                    ctypes.c_uint64]
    data = np.zeros(1024, dtype=np.uint8)
    assert not lib.rng_prover_init(
        data.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)), 1024, 8)
    # primitive entry points return RNG_ERR_NO_GPU, not CPU results
    lib.rng_ntt_fr.restype = ctypes.c_int
    lib.rng_ntt_fr.argtypes = [ctypes.c_void_p, U64P, ctypes.c_uint64,
                               ctypes.c_uint64, ctypes.c_int]
    buf = np.zeros(4 * 8, dtype=np.uint64)
    assert lib.rng_ntt_fr(None, ptr(buf), 8, 1, 0) == RNG_ERR_NO_GPU
    lib.rng_msm_g1.restype = ctypes.c_int
    lib.rng_msm_g1.argtypes = [ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P,
                               ctypes.c_int]
    out = np.zeros(9, dtype=np.uint64)
    assert lib.rng_msm_g1(None, ptr(buf), ptr(buf), 2, ptr(out), 0) == RNG_ERR_NO_GPU


def test_require_gpu_raises(plib):
    if plib.gpu_available:
        pytest.skip("GPU present")
    with pytest.raises(Exception):
        plib.require_gpu()
