"""External prover service daemon (the reference's process-level seam).

An unmodified relayer selects an external prover with --prover-service-url /
--prover-service-password (config/src/cli.rs:55-62; client routes in
proof-manager/src/implementations/external_proof_manager/
prover_service_client.rs:100-147, request/response types api_types.rs:80-135).

This daemon serves ALL 20 prover routes of prover_service_client.rs on top
of the MI355X C ABI (the private-settlement response carries the 4 link
proofs; validity/settlement responses carry their link hint), plus
GET /health.

Wire encoding note: the reference's request bodies are serde
serializations of the Rust witness/statement structs.  This round encodes
witness/statement as flat decimal-string scalar arrays in the reference
struct field order, and proofs/link-proofs/hints in the rkyv field order
(plonk_proof_def.rs) with decimal-string limbs; matching serde's nested JSON
byte-for-byte is scheduled work.  Authentication mirrors the client's bearer
password header.

Run:  python -m renegade_amd.prover_service --port 8000 [--srs-power 15]
"""
import argparse
import ctypes
import os
import threading
import time
from pathlib import Path

import numpy as np

U64P = ctypes.POINTER(ctypes.c_uint64)
REPO = Path(__file__).resolve().parent.parent


def ptr(a):
    return a.ctypes.data_as(U64P)


def scalars_to_json(arr: np.ndarray):
    """4-limb LE records -> decimal strings."""
    a = arr.reshape(-1, 4)
    return [str(int(r[0]) | int(r[1]) << 64 | int(r[2]) << 128 | int(r[3]) << 192)
            for r in a]


def json_to_scalars(vals, count):
    out = np.zeros(4 * count, dtype=np.uint64)
    if len(vals) != count:
        raise ValueError(f"expected {count} scalars, got {len(vals)}")
    for i, v in enumerate(vals):
        x = int(v)
        for k in range(4):
            out[4 * i + k] = (x >> (64 * k)) & 0xFFFFFFFFFFFFFFFF
    return out


def body_scalars(body, key, kind, count, which):
    """Decode body[key] into the flat 4-limb scalar array.

    Two accepted encodings:
    - nested JSON object in the reference's api_types.rs struct shape
      (field names/nesting per renegade_amd.api_schema — the serde layout of
      the in-repo Rust witness/statement structs), scalars as decimal strings;
    - flat decimal-scalar array in struct field order (legacy/debug form).
    """
    from renegade_amd import api_schema
    v = body.get(key)
    if v is None:
        raise ValueError(f"missing {key}")
    if isinstance(v, dict):
        schema = api_schema.SCHEMAS[kind][0 if which == "witness" else 1]
        try:
            flat = api_schema.flatten(v, schema, [])
        except (ValueError, TypeError, KeyError) as e:
            raise ValueError(f"bad {key} shape: {e}")
        if len(flat) != count:
            raise ValueError(f"{key}: schema flattened to {len(flat)}, "
                             f"expected {count}")
        # nested bodies carry canonical values; the ABI takes Montgomery form
        return json_to_scalars([api_schema.to_mont(x) for x in flat], count)
    return json_to_scalars(v, count)


class CohortBatcher:
    """Groups concurrent same-circuit prove requests into one
    `rng_prove_cohort` call (the reference's proof manager drains its job
    queue into a rayon pool — native_proof_manager.rs:193-198; the cohort
    prover is that shape with fused per-round GPU batches).

    Requests wait up to `window_ms` for co-batching; a burst of k same-kind
    requests is proven as one k-cohort.  Single requests degrade to k=1
    (identical bytes to rng_prove)."""

    def __init__(self, svc, window_ms=3.0, max_k=32):
        self.svc = svc
        self.window = window_ms / 1e3
        self.max_k = max_k
        self._pending = {}
        self._lock = threading.Lock()
        self._wake = threading.Event()
        #: telemetry: cohort calls vs proofs served (tests assert batching)
        self.cohort_calls = 0
        self.proofs_served = 0
        svc.lib.rng_prove_cohort.restype = ctypes.c_int
        svc.lib.rng_prove_cohort.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                             ctypes.c_uint64, U64P, U64P, U64P,
                                             U64P, U64P]
        t = threading.Thread(target=self._loop, daemon=True,
                             name="cohort-batcher")
        t.start()

    def prove(self, circuit_name, wires, pubs, seed, n):
        item = {"wires": wires, "pubs": pubs, "seed": seed,
                "done": threading.Event(), "proof": None, "hint": None,
                "err": None}
        with self._lock:
            self._pending.setdefault(circuit_name, []).append(item)
        self._wake.set()
        item["done"].wait()
        if item["err"] is not None:
            raise RuntimeError(item["err"])
        return item["proof"], item["hint"]

    def _loop(self):
        while True:
            self._wake.wait()
            time.sleep(self.window)  # let a burst accumulate
            with self._lock:
                # clear BEFORE swapping: an enqueue that lands after the swap
                # re-sets the event and is drained next loop (clearing after
                # could strand an item whose set() raced the clear)
                self._wake.clear()
                drained = self._pending
                self._pending = {}
            for name, items in drained.items():
                for lo in range(0, len(items), self.max_k):
                    self._run(name, items[lo:lo + self.max_k])

    def _run(self, name, items):
        try:
            svc = self.svc
            pk = svc.pks[name]
            n = int(svc.pk_meta[name]["n"])
            k = len(items)
            wires_all = np.concatenate([it["wires"] for it in items])
            pubs_all = np.concatenate([it["pubs"] for it in items])
            seeds = np.array([it["seed"] for it in items], dtype=np.uint64)
            proofs = np.zeros(157 * k, dtype=np.uint64)
            hlen = 4 * (n + 2) + 9
            hints = np.zeros(hlen * k, dtype=np.uint64)
            rc = svc.lib.rng_prove_cohort(svc.ctx.h, ctypes.c_void_p(pk), k,
                                          ptr(wires_all), ptr(pubs_all),
                                          ptr(seeds), ptr(proofs), ptr(hints))
            if rc != 0:
                raise RuntimeError(f"rng_prove_cohort rc={rc}")
            self.cohort_calls += 1
            self.proofs_served += k
            for i, it in enumerate(items):
                it["proof"] = proofs[157 * i:157 * (i + 1)].copy()
                it["hint"] = hints[hlen * i:hlen * (i + 1)].copy()
        except Exception as e:  # noqa: BLE001 — fail every waiter loudly
            for it in items:
                it["err"] = str(e)
        finally:
            for it in items:
                it["done"].set()


class ProverService:
    """Holds the GPU context + per-circuit PKs (preprocessed once, shared
    read-only across request threads like the reference's key cache,
    traits.rs:80-92)."""

    #: (route cache name, builder, builder args) preloaded by --preload;
    #: route names match prove_kind's `path.lstrip('/')` keys
    PRELOAD = [
        ("valid_balance_create", "rng_circ_build_vbc", ()),
        ("intent_and_balance_private_settlement", "rng_circ_build_settlement", ()),
        ("prove-valid-deposit", "rng_circ_build_valid_deposit", ()),
        ("prove-valid-withdrawal", "rng_circ_build_valid_withdrawal", ()),
        ("prove-valid-order-cancellation", "rng_circ_build_valid_order_cancellation", ()),
        ("prove-intent-and-balance-validity", "rng_circ_build_validity", (0,)),
        ("prove-intent-and-balance-first-fill-validity", "rng_circ_build_ff_validity", (0,)),
        ("prove-intent-only-validity", "rng_circ_build_io_validity", ()),
        ("prove-intent-only-first-fill-validity", "rng_circ_build_ioff", ()),
        ("prove-new-output-balance-validity", "rng_circ_build_nob_validity", ()),
        ("prove-output-balance-validity", "rng_circ_build_ob_validity", (0,)),
        ("prove-intent-and-balance-public-settlement", "rng_circ_build_public_settlement", ()),
        ("prove-intent-and-balance-bounded-settlement", "rng_circ_build_ib_bounded_settlement", ()),
        ("prove-intent-only-public-settlement", "rng_circ_build_io_settlement", ()),
        ("prove-intent-only-bounded-settlement", "rng_circ_build_io_bounded_settlement", ()),
        ("prove-valid-note-redemption", "rng_circ_build_note_redemption", ()),
        ("prove-valid-public-relayer-fee-payment", "rng_circ_build_fee_public_relayer", ()),
        ("prove-valid-public-protocol-fee-payment", "rng_circ_build_fee_public_protocol", ()),
        ("prove-valid-private-relayer-fee-payment", "rng_circ_build_fee_private_relayer", ()),
        ("prove-valid-private-protocol-fee-payment", "rng_circ_build_fee_private_protocol", ()),
    ]

    def preload_keys(self):
        """Preprocess every circuit's PK up front under its route's cache
        name (all 20 in <1 s on the GPU; the reference's CPU startup warmup
        is minutes-scale)."""
        import time
        t0 = time.perf_counter()
        for route_name, builder, extra in self.PRELOAD:
            fn = getattr(self.lib, builder)
            fn.restype = ctypes.c_void_p
            fn.argtypes = [ctypes.c_uint64] * (1 + len(extra))
            h = fn(42, *extra)
            assert h, builder
            self._preprocess(route_name, h)
        print(f"preloaded {len(self.PRELOAD)} circuit PKs in "
              f"{time.perf_counter() - t0:.2f} s", flush=True)

    def _preprocess(self, circuit_name, handle):
        """Preprocess-or-cache the PK for `handle` under `circuit_name`."""
        lib = self.lib
        n = lib.rng_circ_n(handle)
        npub = lib.rng_circ_npub(handle)
        nlg = lib.rng_circ_num_link_groups(handle)
        lg = np.zeros(3 * max(1, nlg), dtype=np.uint64)
        lib.rng_circ_link_groups(handle, ptr(lg))
        sel = np.zeros(13 * n * 4, dtype=np.uint64)
        sigma = np.zeros(5 * n, dtype=np.uint64)
        wires = np.zeros(5 * n * 4, dtype=np.uint64)
        pubs = np.zeros(max(1, npub * 4), dtype=np.uint64)
        lib.rng_circ_get(handle, ptr(sel), ptr(sigma), ptr(wires), ptr(pubs))
        lib.rng_circ_free(handle)
        with self.lock:
            if circuit_name not in self.pks:
                desc = self._Desc(n, npub, ptr(sel), ptr(sigma), 0, None)
                pk = lib.rng_preprocess(self.ctx.h, ctypes.byref(desc))
                if not pk:
                    raise RuntimeError("preprocess failed")
                self.pks[circuit_name] = pk
                self.pk_meta[circuit_name] = dict(n=n, lg=lg.reshape(-1, 3))
        return n, npub, wires, pubs

    def __init__(self, srs_power=15, srs_seed=42, ptau_path=None,
                 batch_window_ms=None):
        import sys
        sys.path.insert(0, str(REPO))
        from renegade_amd import load_prover
        self.plib = load_prover()
        self.plib.require_gpu()
        self.lib = self.plib.lib
        self._sig()
        if ptau_path:
            # production path: the deployment's real ptau bytes (same
            # snarkjs layout srs.rs:63-214 parses)
            ptau = open(ptau_path, "rb").read()
        else:
            # DEV/TEST fallback: deterministic generated SRS from the PRODUCT
            # library (rng_srs_gen_test_ptau) — the daemon never loads oracle
            # code; production deployments pass --ptau with the ceremony file
            self.lib.rng_srs_test_ptau_size.restype = ctypes.c_uint64
            self.lib.rng_srs_test_ptau_size.argtypes = [ctypes.c_int]
            self.lib.rng_srs_gen_test_ptau.argtypes = [
                ctypes.c_int, ctypes.c_uint64, ctypes.POINTER(ctypes.c_uint8),
                ctypes.c_size_t]
            size = self.lib.rng_srs_test_ptau_size(srs_power)
            if not size:
                raise ValueError(f"bad srs power {srs_power}")
            buf = (ctypes.c_uint8 * size)()
            rc = self.lib.rng_srs_gen_test_ptau(srs_power, srs_seed, buf, size)
            if rc != 0:
                raise RuntimeError(f"rng_srs_gen_test_ptau rc={rc}")
            ptau = bytes(buf)
            print(f"WARNING: using deterministic TEST SRS (power {srs_power}, "
                  f"seed {srs_seed}); pass --ptau for production", flush=True)
        self.max_degree = (1 << srs_power) + 2
        self.ctx = self.plib.init(ptau, self.max_degree)
        self.pks = {}
        self.pk_meta = {}
        self.lock = threading.Lock()
        # service-level cohort batching: co-arriving
        # same-circuit requests prove as one fused cohort
        self.batcher = (CohortBatcher(self, window_ms=batch_window_ms)
                        if batch_window_ms else None)

    def _sig(self):
        lib = self.lib
        for name in ["rng_circ_vbc_from_scalars", "rng_circ_settlement_from_scalars"]:
            fn = getattr(lib, name)
            fn.restype = ctypes.c_void_p
            fn.argtypes = [U64P, U64P]
        lib.rng_circ_from_scalars.restype = ctypes.c_void_p
        lib.rng_circ_from_scalars.argtypes = [ctypes.c_int, U64P, U64P]
        lib.rng_ws_sizes.restype = ctypes.c_int
        lib.rng_ws_sizes.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_uint64),
                                     ctypes.POINTER(ctypes.c_uint64)]
        lib.rng_circ_n.restype = ctypes.c_uint64
        lib.rng_circ_n.argtypes = [ctypes.c_void_p]
        lib.rng_circ_npub.restype = ctypes.c_uint64
        lib.rng_circ_npub.argtypes = [ctypes.c_void_p]
        lib.rng_circ_get.argtypes = [ctypes.c_void_p, U64P, U64P, U64P, U64P]
        lib.rng_circ_num_link_groups.restype = ctypes.c_uint64
        lib.rng_circ_num_link_groups.argtypes = [ctypes.c_void_p]
        lib.rng_circ_link_groups.argtypes = [ctypes.c_void_p, U64P]
        lib.rng_circ_free.argtypes = [ctypes.c_void_p]
        lib.rng_preprocess.restype = ctypes.c_void_p
        lib.rng_preprocess.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        lib.rng_prove.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                                  ctypes.c_uint64, U64P, U64P]
        lib.rng_link_proofs.argtypes = [ctypes.c_void_p, ctypes.c_void_p, U64P, U64P,
                                        ctypes.c_uint64, ctypes.c_uint64,
                                        ctypes.c_uint64, U64P]

    class _Desc(ctypes.Structure):
        _fields_ = [("n", ctypes.c_uint64), ("num_public", ctypes.c_uint64),
                    ("selectors", U64P), ("sigma", U64P),
                    ("num_link_groups", ctypes.c_uint64), ("link_offsets", U64P)]

    def _prove_tables(self, circuit_name, handle, want_hint):
        """Preprocess-or-cache, then prove; returns (proof, hint, meta)."""
        lib = self.lib
        n, npub, wires, pubs = self._preprocess(circuit_name, handle)
        # fresh blinder seed per request from the OS CSPRNG — never derive
        # successive proofs' blinders from a counter (the reference blinds
        # with thread_rng at traits.rs:994)
        seed = int.from_bytes(os.urandom(8), "little")
        if self.batcher is not None:
            proof, hint = self.batcher.prove(circuit_name, wires, pubs, seed,
                                             int(n))
            if not want_hint:
                hint = None
            return proof, hint, self.pk_meta[circuit_name]
        pk = self.pks[circuit_name]
        proof = np.zeros(157, dtype=np.uint64)
        hint = np.zeros(4 * (n + 2) + 9, dtype=np.uint64) if want_hint else None
        rc = lib.rng_prove(self.ctx.h, ctypes.c_void_p(pk), ptr(wires), ptr(pubs),
                           ctypes.c_uint64(seed), ptr(proof),
                           ptr(hint) if want_hint else None)
        if rc != 0:
            raise RuntimeError(f"rng_prove rc={rc}")
        return proof, hint, self.pk_meta[circuit_name]

    # ---- link-proof helpers ----
    @staticmethod
    def _hint_n(h):
        """Domain size encoded by a hint's length (4*(n+2)+9 u64)."""
        return (h.size - 9) // 4 - 2

    @staticmethod
    def _ext_hint(h, n_big):
        out = np.zeros(4 * (n_big + 2) + 9, dtype=np.uint64)
        out[:h.size - 9] = h[:-9]
        out[-9:] = h[-9:]
        return out

    def _pk_for_n(self, n):
        """Any cached proving key whose domain is n (cross-domain links run
        under the larger domain's key; --preload guarantees coverage)."""
        for name, meta in self.pk_meta.items():
            if int(meta["n"]) == n:
                return self.pks[name]
        raise ValueError(f"no cached proving key for domain {n}; "
                         "start the service with --preload")

    def _link(self, my_hint, ext_hint, group, n_mine):
        """Link an external validity hint against this proof's hint at the
        given (alignment, offset, count) group — validity first, matching
        native_proof_manager.rs:746-777."""
        align, off, cnt = (int(x) for x in group)
        n_ext = self._hint_n(ext_hint)
        n_big = max(n_ext, n_mine)
        pk_big = self._pk_for_n(n_big)
        ha = self._ext_hint(ext_hint, n_big)
        hb = self._ext_hint(my_hint, n_big)
        lp = np.zeros(18, dtype=np.uint64)
        rc = self.lib.rng_link_proofs(self.ctx.h, ctypes.c_void_p(pk_big), ptr(ha),
                                      ptr(hb), align, off, cnt, ptr(lp))
        if rc != 0:
            raise ValueError(f"rng_link_proofs rc={rc}")
        return [str(int(x)) for x in lp]

    @staticmethod
    def _hint_from_body(body, key):
        if key not in body:
            raise ValueError(f"missing {key}")
        h = np.array([int(x) for x in body[key]], dtype=np.uint64)
        if (h.size - 9) % 4 or h.size < 9 + 4 * 3:
            raise ValueError(f"bad hint size for {key}")
        return h

    # ---- route handlers ----
    def prove_valid_balance_create(self, body):
        w = body_scalars(body, "witness", 0, 12, "witness")
        s = body_scalars(body, "statement", 0, 13, "statement")
        h = self.lib.rng_circ_vbc_from_scalars(ptr(w), ptr(s))
        if not h:
            raise ValueError("unsatisfied witness/statement")
        proof, _, _ = self._prove_tables("valid_balance_create", h, False)
        # proof = flat 157-u64 buffer in the rkyv field order (plonk_proof_def.rs)
        return {"proof": [str(int(x)) for x in proof]}

    def prove_kind(self, route_name, kind, body, want_hint):
        """Generic route: witness/statement scalar arrays -> proof
        (+ link hint for validity circuits; + link proofs for the public/
        bounded settlement routes whose requests carry validity hints —
        api_types.rs SettlementProofResponse / PublicSettlementProofResponse).
        """
        nw, ns = ctypes.c_uint64(), ctypes.c_uint64()
        if self.lib.rng_ws_sizes(kind, ctypes.byref(nw), ctypes.byref(ns)) != 0:
            raise ValueError(f"unknown kind {kind}")
        w = body_scalars(body, "witness", kind, nw.value, "witness") if nw.value \
            else np.zeros(0, dtype=np.uint64)
        s = body_scalars(body, "statement", kind, ns.value, "statement")
        h = self.lib.rng_circ_from_scalars(kind, ptr(w), ptr(s))
        if not h:
            raise ValueError("unsatisfied witness/statement")
        proof, hint, meta = self._prove_tables(route_name, h, want_hint)
        out = {"proof": [str(int(x)) for x in proof]}
        if kind in (11, 12):  # ib public/bounded settlement: 2 link proofs
            groups = {int(r[2]): r for r in meta["lg"]}
            n_mine = int(meta["n"])
            out["validity_link_proof"] = self._link(
                hint, self._hint_from_body(body, "validity_link_hint"),
                groups[17], n_mine)
            out["output_balance_link_proof"] = self._link(
                hint, self._hint_from_body(body, "output_balance_link_hint"),
                groups[11], n_mine)
        elif kind in (13, 14):  # intent-only settlements: 1 link proof
            out["link_proof"] = self._link(
                hint, self._hint_from_body(body, "validity_link_hint"),
                meta["lg"][0], int(meta["n"]))
        elif want_hint:
            out["link_hint"] = [str(int(x)) for x in hint]
        return out

    def prove_private_settlement(self, body):
        w = body_scalars(body, "witness", 10, 64, "witness")
        s = body_scalars(body, "statement", 10, 17, "statement")
        h = self.lib.rng_circ_settlement_from_scalars(ptr(w), ptr(s))
        if not h:
            raise ValueError("unsatisfied witness/statement")
        proof, hint, meta = self._prove_tables("intent_and_balance_private_settlement",
                                               h, True)
        n_mine = int(meta["n"])
        # 4 link proofs: settlement hint vs the 4 supplied hints, one per
        # link group (party0, party1 validity; party0, party1 output balance;
        # group order = creation order in the circuit).  Real hints come from
        # DIFFERENT domains (intent-and-balance validity n=16384, output-
        # balance validity n=8192), so each link runs cross-domain via _link:
        # zero-extend to the larger domain, prove under that domain's PK
        # (requires --preload; native_proof_manager.rs:746-777).
        names = ["validity_link_proof_0", "validity_link_proof_1",
                 "output_balance_link_proof_0", "output_balance_link_proof_1"]
        group_order = [0, 2, 1, 3]  # groups created party0(v,o), party1(v,o)
        out = {"proof": [str(int(x)) for x in proof]}
        for name, gidx in zip(names, group_order):
            ext_hint = self._hint_from_body(body, name.replace("_proof", "_hint"))
            out[name] = self._link(hint, ext_hint, meta["lg"][gidx], n_mine)
        return out


def create_app(service=None, password=None):
    from fastapi import Body, FastAPI, HTTPException, Request

    # handlers are SYNC (def, not async def): FastAPI then runs them in its
    # thread pool, so concurrent requests actually overlap (ctypes releases
    # the GIL) and co-arriving requests can batch into one cohort
    app = FastAPI(title="renegade_amd prover service")
    svc = service

    def auth(request: Request):
        if password is None:
            return
        import hmac
        hdr = request.headers.get("authorization", "")
        if not hmac.compare_digest(hdr.encode(), f"Bearer {password}".encode()):
            raise HTTPException(status_code=401, detail="bad password")

    @app.get("/health")
    def health():
        return {"status": "ok", "backend": svc.plib.version}

    @app.post("/prove-valid-balance-create")
    def vbc(request: Request, body: dict = Body(...)):
        auth(request)
        try:
            return svc.prove_valid_balance_create(body)
        except ValueError as e:
            raise HTTPException(status_code=400, detail=str(e))

    @app.post("/prove-intent-and-balance-private-settlement")
    def settle(request: Request, body: dict = Body(...)):
        auth(request)
        try:
            return svc.prove_private_settlement(body)
        except ValueError as e:
            raise HTTPException(status_code=400, detail=str(e))

    # the remaining 18 routes of prover_service_client.rs:100-147, served by
    # the generic from-scalars prover; validity/settlement circuits that enter
    # later link proofs also return their link hint
    ROUTES = {
        "/prove-valid-deposit": (1, False),
        "/prove-valid-withdrawal": (2, False),
        "/prove-valid-order-cancellation": (3, False),
        "/prove-intent-and-balance-validity": (4, True),
        "/prove-intent-and-balance-first-fill-validity": (5, True),
        "/prove-intent-only-validity": (6, True),
        "/prove-intent-only-first-fill-validity": (7, True),
        "/prove-new-output-balance-validity": (8, True),
        "/prove-output-balance-validity": (9, True),
        "/prove-intent-and-balance-public-settlement": (11, True),
        "/prove-intent-and-balance-bounded-settlement": (12, True),
        "/prove-intent-only-public-settlement": (13, True),
        "/prove-intent-only-bounded-settlement": (14, True),
        "/prove-valid-note-redemption": (15, False),
        "/prove-valid-public-relayer-fee-payment": (16, False),
        "/prove-valid-public-protocol-fee-payment": (17, False),
        "/prove-valid-private-relayer-fee-payment": (18, False),
        "/prove-valid-private-protocol-fee-payment": (19, False),
    }

    def make_route(path, kind, want_hint):
        def handler(request: Request, body: dict = Body(...)):
            auth(request)
            try:
                return svc.prove_kind(path.lstrip("/"), kind, body, want_hint)
            except ValueError as e:
                raise HTTPException(status_code=400, detail=str(e))
        app.post(path)(handler)

    for path, (kind, want_hint) in ROUTES.items():
        make_route(path, kind, want_hint)

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--srs-power", type=int, default=15)  # covers all 20 circuits (n <= 32768)
    ap.add_argument("--password", default=os.environ.get("PROVER_SERVICE_PASSWORD"))
    ap.add_argument("--preload", action="store_true",
                    help="preprocess all 20 circuit PKs at startup (<1 s)")
    ap.add_argument("--ptau", default=None,
                    help="path to the production ptau file (snarkjs layout); "
                         "omitted = deterministic TEST SRS")
    ap.add_argument("--batch-window-ms", type=float, default=3.0,
                    help="co-batch same-circuit requests arriving within this "
                         "window into one fused proof cohort (0 = off)")
    args = ap.parse_args()
    import uvicorn
    svc = ProverService(srs_power=args.srs_power, ptau_path=args.ptau,
                        batch_window_ms=args.batch_window_ms or None)
    if args.preload:
        svc.preload_keys()
    app = create_app(svc, args.password)
    uvicorn.run(app, host="0.0.0.0", port=args.port)


if __name__ == "__main__":
    main()
