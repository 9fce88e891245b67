#!/usr/bin/env python3
"""Ad-hoc service stress: mixed-circuit request bursts through the cohort
batcher (exercises mixed-n scratch growth + concurrent routes)."""
import ctypes
import sys
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)


def main():
    from fastapi.testclient import TestClient
    from renegade_amd.prover_service import ProverService, create_app, scalars_to_json
    svc = ProverService(srs_power=15, batch_window_ms=4.0)
    svc.preload_keys()
    c = TestClient(create_app(svc, password=None))
    lib = svc.lib
    lib.rng_ws_sizes.restype = ctypes.c_int
    lib.rng_ws_sizes.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_uint64),
                                 ctypes.POINTER(ctypes.c_uint64)]
    lib.rng_witness_statement.restype = ctypes.c_int
    lib.rng_witness_statement.argtypes = [ctypes.c_int, ctypes.c_uint64, U64P, U64P]
    kinds = {3: "/prove-valid-order-cancellation", 4: "/prove-intent-and-balance-validity",
             6: "/prove-intent-only-validity", 9: "/prove-output-balance-validity",
             15: "/prove-valid-note-redemption"}
    bodies = {}
    for kind in kinds:
        nw, ns = ctypes.c_uint64(), ctypes.c_uint64()
        assert lib.rng_ws_sizes(kind, ctypes.byref(nw), ctypes.byref(ns)) == 0
        w = np.zeros(4 * nw.value, dtype=np.uint64)
        s = np.zeros(4 * ns.value, dtype=np.uint64)
        assert lib.rng_witness_statement(kind, 5, ptr(w), ptr(s)) == 0
        bodies[kind] = {"witness": scalars_to_json(w), "statement": scalars_to_json(s)}

    def hit(i):
        kind = list(kinds)[i % len(kinds)]
        r = c.post(kinds[kind], json=bodies[kind])
        assert r.status_code == 200, f"{kinds[kind]}: {r.text}"
        return len(r.json()["proof"])

    N = 120
    with ThreadPoolExecutor(max_workers=24) as ex:
        res = list(ex.map(hit, range(N)))
    assert all(x == 157 for x in res)
    print(f"stress ok: {N} mixed-kind requests, {svc.batcher.cohort_calls} cohort "
          f"calls, {svc.batcher.proofs_served} proofs served", flush=True)


if __name__ == "__main__":
    main()
