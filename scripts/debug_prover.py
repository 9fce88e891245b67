import ctypes, faulthandler, sys
faulthandler.enable()
sys.path.insert(0, "/root/repo")
import numpy as np
from renegade_amd import load_prover
from tests.orc_bindings import OracleLib

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

plib = load_prover(); lib = plib.lib
orc = OracleLib("/root/repo/oracle/liborc.so")
lib.rng_testcirc_build.restype = ctypes.c_void_p
lib.rng_testcirc_build.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
lib.rng_circ_n.restype = ctypes.c_uint64; lib.rng_circ_n.argtypes=[ctypes.c_void_p]
lib.rng_circ_npub.restype = ctypes.c_uint64; lib.rng_circ_npub.argtypes=[ctypes.c_void_p]
lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
lib.rng_preprocess.restype = ctypes.c_void_p
lib.rng_preprocess.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
lib.rng_prove.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P, ctypes.c_uint64, U64P, U64P]

print("build circuit", flush=True)
h = lib.rng_testcirc_build(777, 6); assert h
n = lib.rng_circ_n(h); npub = lib.rng_circ_npub(h)
print("n", n, "npub", npub, flush=True)
sel = np.zeros(13*n*4, dtype=np.uint64); sigma = np.zeros(5*n, dtype=np.uint64)
wires = np.zeros(5*n*4, dtype=np.uint64); pubs = np.zeros(max(1,npub*4), dtype=np.uint64)
lib.rng_circ_get(h, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
print("got tables", flush=True)
power = max(4, int(n).bit_length())
ptau = orc.srs_generate_ptau(power, seed=42)
print("srs", power, flush=True)
ctx = plib.init(ptau, (1<<power)+2)
print("ctx ok", flush=True)

class Desc(ctypes.Structure):
    _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                ("selectors", U64P), ("sigma", U64P),
                ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]
desc = Desc(n, npub, ptr(sel), ptr(sigma), 0, None)
pk = lib.rng_preprocess(ctx.h, ctypes.byref(desc))
print("preprocess ->", pk, flush=True)
proof = np.zeros(157, dtype=np.uint64)
rc = lib.rng_prove(ctx.h, ctypes.c_void_p(pk), ptr(wires), ptr(pubs), 9, ptr(proof), None)
print("prove rc", rc, flush=True)
print("proof head", proof[:4], flush=True)
