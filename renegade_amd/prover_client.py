"""Python client for the prover-service daemon — mirrors the reference
relayer's `ProverServiceClient` (proof-manager/src/implementations/
external_proof_manager/prover_service_client.rs:100-147): one method per
route, bearer-password auth, witness/statement as scalar lists.

Scalars are Montgomery-limb values encoded as decimal strings (the same
encoding `prover_service.scalars_to_json` emits); helpers accept numpy
4-limb arrays directly.

    client = ProverServiceClient("http://gpu-box:8000", password="...")
    resp = client.prove_valid_deposit(witness_scalars, statement_scalars)
    proof = resp["proof"]          # 157 u64 decimal strings (rkyv order)
"""
from __future__ import annotations

import numpy as np


def scalars_to_json(arr):
    a = np.asarray(arr, dtype=np.uint64).reshape(-1, 4)
    return [str(int(r[0]) | int(r[1]) << 64 | int(r[2]) << 128 | int(r[3]) << 192)
            for r in a]


def limbs_to_json(arr):
    """Flat u64 buffers (hints, proofs) -> decimal strings."""
    return [str(int(x)) for x in np.asarray(arr, dtype=np.uint64)]


#: route -> (method name, needs hints)
ROUTES = {
    "prove_valid_balance_create": ("/prove-valid-balance-create", ()),
    "prove_valid_deposit": ("/prove-valid-deposit", ()),
    "prove_valid_order_cancellation": ("/prove-valid-order-cancellation", ()),
    "prove_valid_withdrawal": ("/prove-valid-withdrawal", ()),
    "prove_intent_and_balance_validity": ("/prove-intent-and-balance-validity", ()),
    "prove_intent_and_balance_first_fill_validity":
        ("/prove-intent-and-balance-first-fill-validity", ()),
    "prove_intent_only_validity": ("/prove-intent-only-validity", ()),
    "prove_intent_only_first_fill_validity":
        ("/prove-intent-only-first-fill-validity", ()),
    "prove_new_output_balance_validity": ("/prove-new-output-balance-validity", ()),
    "prove_output_balance_validity": ("/prove-output-balance-validity", ()),
    "prove_intent_and_balance_bounded_settlement":
        ("/prove-intent-and-balance-bounded-settlement",
         ("validity_link_hint", "output_balance_link_hint")),
    "prove_intent_and_balance_private_settlement":
        ("/prove-intent-and-balance-private-settlement",
         ("validity_link_hint_0", "validity_link_hint_1",
          "output_balance_link_hint_0", "output_balance_link_hint_1")),
    "prove_intent_and_balance_public_settlement":
        ("/prove-intent-and-balance-public-settlement",
         ("validity_link_hint", "output_balance_link_hint")),
    "prove_intent_only_bounded_settlement":
        ("/prove-intent-only-bounded-settlement", ("validity_link_hint",)),
    "prove_intent_only_public_settlement":
        ("/prove-intent-only-public-settlement", ("validity_link_hint",)),
    "prove_valid_note_redemption": ("/prove-valid-note-redemption", ()),
    "prove_valid_private_protocol_fee_payment":
        ("/prove-valid-private-protocol-fee-payment", ()),
    "prove_valid_private_relayer_fee_payment":
        ("/prove-valid-private-relayer-fee-payment", ()),
    "prove_valid_public_protocol_fee_payment":
        ("/prove-valid-public-protocol-fee-payment", ()),
    "prove_valid_public_relayer_fee_payment":
        ("/prove-valid-public-relayer-fee-payment", ()),
}


class ProverServiceClient:
    """One method per reference route; `transport` defaults to `requests`
    but anything with a `.post(url, json=..., headers=...)` works (the tests
    pass FastAPI's TestClient)."""

    def __init__(self, url, password=None, transport=None, timeout=120):
        self.url = url.rstrip("/")
        self.password = password
        self.timeout = timeout
        if transport is None:
            import requests
            transport = requests
        self.transport = transport

    def _headers(self):
        h = {}
        if self.password is not None:
            h["authorization"] = f"Bearer {self.password}"
        return h

    def health(self):
        r = self.transport.get(f"{self.url}/health")
        r.raise_for_status()
        return r.json()

    def _post(self, path, body):
        kwargs = dict(json=body, headers=self._headers())
        try:  # requests wants timeout; TestClient does not accept it
            r = self.transport.post(f"{self.url}{path}", timeout=self.timeout, **kwargs)
        except TypeError:
            r = self.transport.post(f"{self.url}{path}", **kwargs)
        if r.status_code != 200:
            raise RuntimeError(f"{path}: HTTP {r.status_code}: {r.text}")
        return r.json()


def _make_method(path, hint_keys):
    def method(self, witness, statement, *hints):
        if len(hints) != len(hint_keys):
            raise TypeError(f"{path} expects {len(hint_keys)} link hint(s), "
                            f"got {len(hints)}")
        body = {"witness": scalars_to_json(witness),
                "statement": scalars_to_json(statement)}
        for key, hint in zip(hint_keys, hints):
            body[key] = limbs_to_json(hint)
        return self._post(path, body)
    method.__doc__ = f"POST {path} (hints: {list(hint_keys) or 'none'})"
    return method


for _name, (_path, _hints) in ROUTES.items():
    setattr(ProverServiceClient, _name, _make_method(_path, _hints))
