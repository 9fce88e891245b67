"""api_types.rs request-shape schemas: sizes match the ABI's per-kind scalar
counts, and nested<->flat conversion round-trips the fixed test vectors for
every circuit kind.  CPU-only (the generators are host code in the prover
library)."""
import ctypes
from pathlib import Path

import numpy as np
import pytest

from renegade_amd import api_schema
from renegade_amd.api_schema import SCHEMAS, flatten, schema_size, unflatten

REPO = Path(__file__).resolve().parent.parent
U64P = ctypes.POINTER(ctypes.c_uint64)
ptr = lambda a: a.ctypes.data_as(U64P)

#: dedicated-route kinds -> (nw, ns); others come from rng_ws_sizes
DEDICATED = {0: (12, 13), 10: (64, 17)}


@pytest.fixture(scope="module")
def lib():
    lib = ctypes.CDLL(str(REPO / "renegade_amd" / "librenegade_prover.so"))
    lib.rng_ws_sizes.restype = ctypes.c_int
    lib.rng_ws_sizes.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_uint64),
                                 ctypes.POINTER(ctypes.c_uint64)]
    lib.rng_witness_statement.restype = ctypes.c_int
    lib.rng_witness_statement.argtypes = [ctypes.c_int, ctypes.c_uint64, U64P, U64P]
    lib.rng_vbc_witness_statement.argtypes = [ctypes.c_uint64, U64P, U64P]
    lib.rng_witness_statement_party.restype = ctypes.c_int
    lib.rng_witness_statement_party.argtypes = [ctypes.c_int, ctypes.c_uint64,
                                                ctypes.c_uint64, U64P, U64P]
    return lib


def _sizes(lib, kind):
    if kind in DEDICATED:
        return DEDICATED[kind]
    nw, ns = ctypes.c_uint64(), ctypes.c_uint64()
    assert lib.rng_ws_sizes(kind, ctypes.byref(nw), ctypes.byref(ns)) == 0
    return nw.value, ns.value


def _vectors(lib, kind, seed=11):
    nw, ns = _sizes(lib, kind)
    w = np.zeros(4 * nw, dtype=np.uint64)
    s = np.zeros(4 * ns, dtype=np.uint64)
    if kind == 0:
        lib.rng_vbc_witness_statement(seed, ptr(w), ptr(s))
    elif kind == 10:
        assert lib.rng_witness_statement_party(10, seed, 0, ptr(w), ptr(s)) == 0
    else:
        assert lib.rng_witness_statement(kind, seed, ptr(w), ptr(s)) == 0
    return w.reshape(-1, 4), s.reshape(-1, 4)


def _to_ints(limbs):
    return [int(r[0]) | int(r[1]) << 64 | int(r[2]) << 128 | int(r[3]) << 192
            for r in limbs]


class TestApiSchema:
    def test_sizes_match_abi(self, lib):
        for kind, (wsch, ssch) in SCHEMAS.items():
            nw, ns = _sizes(lib, kind)
            assert schema_size(wsch) == nw, f"kind {kind} witness"
            assert schema_size(ssch) == ns, f"kind {kind} statement"

    @pytest.mark.parametrize("kind", sorted(SCHEMAS))
    def test_round_trip_vectors(self, lib, kind):
        """Montgomery flat -> canonical nested (api_types struct shape) ->
        Montgomery flat is the identity on every kind's test vectors."""
        w, s = _vectors(lib, kind)
        for limbs, schema in [(w, SCHEMAS[kind][0]), (s, SCHEMAS[kind][1])]:
            flat = _to_ints(limbs)
            canon = [api_schema.from_mont(x) for x in flat]
            nested, pos = unflatten(canon, schema)
            assert pos == len(flat)
            again = flatten(nested, schema, [])
            assert [api_schema.to_mont(x) for x in again] == flat
            # u64 leaves must actually be small in canonical form (layout check)
            assert all(0 <= x < (1 << 256) for x in canon)

    def test_nested_types(self, lib):
        """u64 leaves surface as JSON ints, bools as bools, scalars as decimal
        strings — the serde-natural encodings of the in-repo struct fields."""
        w, _ = _vectors(lib, 4)  # intent-and-balance validity
        nested, _ = unflatten([api_schema.from_mont(x) for x in _to_ints(w)],
                              SCHEMAS[4][0])
        assert isinstance(nested["old_intent"]["recovery_stream"]["index"], int)
        assert isinstance(nested["old_intent_opening"]["indices"][0], bool)
        assert isinstance(nested["intent"]["in_token"], str)
        assert isinstance(nested["intent"]["amount_in"], int)
        assert set(nested["balance"].keys()) == {
            "mint", "owner", "relayer_fee_recipient", "authority",
            "relayer_fee_balance", "protocol_fee_balance", "amount"}
        assert set(nested["balance"]["authority"].keys()) == {"point"}

    def test_body_scalars_accepts_both(self, lib):
        from renegade_amd.prover_service import body_scalars, scalars_to_json
        w, _ = _vectors(lib, 13)  # intent-only settlement (5 scalars)
        flat_json = scalars_to_json(w)
        nested, _ = unflatten([api_schema.from_mont(int(x)) for x in flat_json],
                              SCHEMAS[13][0])
        a = body_scalars({"witness": flat_json}, "witness", 13, 5, "witness")
        b = body_scalars({"witness": nested}, "witness", 13, 5, "witness")
        assert np.array_equal(a, b)
        with pytest.raises(ValueError):
            bad = dict(nested)
            del bad["intent"]
            body_scalars({"witness": bad}, "witness", 13, 5, "witness")
