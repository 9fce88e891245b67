"""Nested request-body schemas mirroring the reference's api_types.rs.

The prover-service client POSTs serde JSON of the Rust witness/statement
structs (api_types.rs:140-345).  The struct field names and nesting are
IN-REPO (crates/circuits/circuits-core/src/zk_circuits/**, circuit-types
primitives), so this module pins them: each kind's schema lists the fields
in declaration order, and flattening a nested body yields exactly the flat
scalar order `rng_circ_from_scalars` consumes (which follows the same struct
order).

What is NOT pinnable offline (documented): the serde encoding
of `Scalar` itself (ark-mpc GenericScalar, a non-vendored git dependency) and
of the jf-plonk Proof/ProofLinkingHint types.  This module encodes Scalar
leaves as decimal strings, u64 leaves (Amount, index, block_deadline) as JSON
integers, and bool leaves as JSON booleans; hints/proofs stay flat arrays.

Leaf codes: "s" Scalar, "u" u64, "b" bool.
Node forms:  ("struct", [(name, schema), ...]) | ("list", schema, count).
"""

MERKLE_HEIGHT = 10

S, U, B = "s", "u", "b"

# BN254 scalar field + the Montgomery radix.  The FLAT wire/ABI format
# carries Montgomery-form limb values (the internal Fr representation, as
# plonk_proof_def.rs:22-52 pins for proof records); nested api_types bodies
# carry CANONICAL values (serde-natural: `amount_in: 1000`), converted at
# this boundary.
FR_MOD = 0x30644E72E131A029B85045B68181585D2833E84879B97091_43E1F593F0000001
_R = (1 << 256) % FR_MOD
_R_INV = pow(_R, -1, FR_MOD)


def to_mont(x):
    return (int(x) % FR_MOD) * _R % FR_MOD


def from_mont(x):
    return int(x) * _R_INV % FR_MOD


def struct(*fields):
    return ("struct", list(fields))


def lst(schema, n):
    return ("list", schema, n)


# --- primitives (circuit-types/src/primitives/*) ---
FIXED_POINT = struct(("repr", S))                       # fixed_point.rs:89
JUBJUB_PT = struct(("x", S), ("y", S))                  # BabyJubJubPoint
SCHNORR_PK = struct(("point", JUBJUB_PT))               # schnorr.rs
SCHNORR_SIG = struct(("s", S), ("r", JUBJUB_PT))        # schnorr.rs
CSPRNG = struct(("seed", S), ("index", U))              # PoseidonCSPRNG
MERKLE_OPENING = struct(("elems", lst(S, MERKLE_HEIGHT)),
                        ("indices", lst(B, MERKLE_HEIGHT)))  # merkle.rs
PARTIAL_COMMITMENT = struct(("private_commitment", S),
                            ("partial_public_commitment", S))

# --- darkpool types (field order per the reference structs) ---
OBLIGATION = struct(("input_token", S), ("output_token", S),
                    ("amount_in", U), ("amount_out", U))  # settlement_obligation.rs
INTENT = struct(("in_token", S), ("out_token", S), ("owner", S),
                ("min_price", FIXED_POINT), ("amount_in", U))  # intent.rs:49-70
BALANCE = struct(("mint", S), ("owner", S), ("relayer_fee_recipient", S),
                 ("authority", SCHNORR_PK), ("relayer_fee_balance", U),
                 ("protocol_fee_balance", U), ("amount", U))  # balance.rs
PMS = struct(("relayer_fee_balance", S), ("protocol_fee_balance", S),
             ("amount", S))  # PostMatchBalanceShare
INTENT_SHARE = struct(("in_token", S), ("out_token", S), ("owner", S),
                      ("min_price", struct(("repr", S))), ("amount_in", S))
PRE_MATCH_INTENT_SHARE = struct(("in_token", S), ("out_token", S), ("owner", S),
                                ("min_price", struct(("repr", S))))
BALANCE_SHARE = struct(("mint", S), ("owner", S), ("relayer_fee_recipient", S),
                       ("authority", struct(("point", JUBJUB_PT))),
                       ("relayer_fee_balance", S), ("protocol_fee_balance", S),
                       ("amount", S))  # DarkpoolBalanceShare
PRE_MATCH_BALANCE_SHARE = struct(("mint", S), ("owner", S),
                                 ("relayer_fee_recipient", S),
                                 ("authority", struct(("point", JUBJUB_PT))))
NOTE = struct(("mint", S), ("amount", U), ("receiver", S), ("blinder", S))
DEPOSIT = struct(("from", S), ("token", S), ("amount", U))
WITHDRAWAL = struct(("to", S), ("token", S), ("amount", U))
BMR = struct(("internal_party_input_token", S), ("internal_party_output_token", S),
             ("min_internal_party_amount_in", U), ("max_internal_party_amount_in", U),
             ("price", FIXED_POINT), ("block_deadline", U))  # BoundedMatchResult
FEE_RATES = struct(("relayer_fee_rate", FIXED_POINT),
                   ("protocol_fee_rate", FIXED_POINT))
ELGAMAL_CT3 = struct(("ephemeral_key", JUBJUB_PT), ("ciphertext", lst(S, 3)))


def _state_wrapper(inner, share):
    # StateWrapper<T> (state_wrapper.rs): recovery_stream, share_stream,
    # inner, public_share — the same order the flat layout uses
    return struct(("recovery_stream", CSPRNG), ("share_stream", CSPRNG),
                  ("inner", inner), ("public_share", share))


STATE_INTENT = _state_wrapper(INTENT, INTENT_SHARE)      # DarkpoolStateIntent
STATE_BALANCE = _state_wrapper(BALANCE, BALANCE_SHARE)   # DarkpoolStateBalance


def _settlement_party(i):
    return [
        (f"settlement_obligation{i}", OBLIGATION),
        (f"intent{i}", INTENT),
        (f"pre_settlement_amount_public_share{i}", S),
        (f"input_balance{i}", BALANCE),
        (f"pre_settlement_in_balance_shares{i}", PMS),
        (f"output_balance{i}", BALANCE),
        (f"pre_settlement_out_balance_shares{i}", PMS),
    ]


_PUB_SETTLE_WITNESS = struct(
    ("intent", INTENT), ("pre_settlement_amount_public_share", S),
    ("in_balance", BALANCE), ("pre_settlement_in_balance_shares", PMS),
    ("out_balance", BALANCE), ("pre_settlement_out_balance_shares", PMS))

_FEE_PUBLIC_STATEMENT_TAIL = [
    ("merkle_root", S), ("old_balance_nullifier", S),
    ("new_balance_commitment", S), ("recovery_id", S)]

#: kind -> (witness_schema, statement_schema).  Kind 0 = VALID BALANCE CREATE,
#: kind 10 = INTENT AND BALANCE PRIVATE SETTLEMENT (the dedicated routes).
SCHEMAS = {
    0: (struct(("initial_share_stream", CSPRNG), ("initial_recovery_stream", CSPRNG),
               ("balance", BALANCE)),
        struct(("deposit", DEPOSIT), ("balance_commitment", S), ("recovery_id", S),
               ("new_balance_share", BALANCE_SHARE))),
    1: (struct(("old_balance", STATE_BALANCE), ("old_balance_opening", MERKLE_OPENING)),
        struct(("deposit", DEPOSIT), ("merkle_root", S), ("old_balance_nullifier", S),
               ("new_balance_commitment", S), ("recovery_id", S),
               ("new_amount_share", S))),
    2: (struct(("old_balance", STATE_BALANCE), ("old_balance_opening", MERKLE_OPENING)),
        struct(("withdrawal", WITHDRAWAL), ("merkle_root", S),
               ("old_balance_nullifier", S), ("new_balance_commitment", S),
               ("recovery_id", S), ("new_amount_share", S))),
    3: (struct(("old_intent", STATE_INTENT), ("old_intent_opening", MERKLE_OPENING)),
        struct(("merkle_root", S), ("old_intent_nullifier", S), ("owner", S))),
    4: (struct(("old_intent", STATE_INTENT), ("old_intent_opening", MERKLE_OPENING),
               ("intent", INTENT), ("new_amount_public_share", S),
               ("old_balance", STATE_BALANCE),
               ("old_balance_opening", MERKLE_OPENING), ("balance", BALANCE),
               ("post_match_balance_shares", PMS)),
        struct(("intent_merkle_root", S), ("old_intent_nullifier", S),
               ("new_intent_partial_commitment", PARTIAL_COMMITMENT),
               ("intent_recovery_id", S), ("balance_merkle_root", S),
               ("old_balance_nullifier", S),
               ("new_balance_partial_commitment", PARTIAL_COMMITMENT),
               ("balance_recovery_id", S))),
    5: (struct(("intent", INTENT), ("initial_intent_share_stream", CSPRNG),
               ("initial_intent_recovery_stream", CSPRNG),
               ("private_intent_shares", INTENT_SHARE),
               ("new_amount_public_share", S),
               ("intent_authorization_signature", SCHNORR_SIG),
               ("old_balance", STATE_BALANCE), ("balance", BALANCE),
               ("post_match_balance_shares", PMS),
               ("old_balance_opening", MERKLE_OPENING)),
        struct(("merkle_root", S), ("intent_public_share", PRE_MATCH_INTENT_SHARE),
               ("intent_private_share_commitment", S), ("intent_recovery_id", S),
               ("balance_partial_commitment", PARTIAL_COMMITMENT),
               ("old_balance_nullifier", S), ("balance_recovery_id", S))),
    6: (struct(("old_intent", STATE_INTENT), ("old_intent_opening", MERKLE_OPENING),
               ("intent", INTENT)),
        struct(("owner", S), ("merkle_root", S), ("old_intent_nullifier", S),
               ("new_amount_public_share", S),
               ("new_intent_partial_commitment", PARTIAL_COMMITMENT),
               ("recovery_id", S))),
    7: (struct(("intent", INTENT), ("initial_intent_share_stream", CSPRNG),
               ("initial_intent_recovery_stream", CSPRNG),
               ("private_shares", INTENT_SHARE)),
        struct(("owner", S), ("intent_private_commitment", S), ("recovery_id", S),
               ("intent_public_share", INTENT_SHARE))),
    8: (struct(("new_balance", STATE_BALANCE), ("balance", BALANCE),
               ("post_match_balance_shares", PMS),
               ("existing_balance", STATE_BALANCE),
               ("existing_balance_opening", MERKLE_OPENING),
               ("new_balance_authorization_signature", SCHNORR_SIG)),
        struct(("existing_balance_merkle_root", S),
               ("existing_balance_nullifier", S),
               ("pre_match_balance_shares", PRE_MATCH_BALANCE_SHARE),
               ("new_balance_partial_commitment", PARTIAL_COMMITMENT),
               ("recovery_id", S))),
    9: (struct(("old_balance", STATE_BALANCE), ("balance_opening", MERKLE_OPENING),
               ("balance", BALANCE), ("post_match_balance_shares", PMS)),
        struct(("merkle_root", S), ("old_balance_nullifier", S),
               ("new_partial_commitment", PARTIAL_COMMITMENT), ("recovery_id", S))),
    10: (struct(*(_settlement_party(0) + _settlement_party(1))),
         struct(("new_amount_public_share0", S),
                ("new_in_balance_public_shares0", PMS),
                ("new_out_balance_public_shares0", PMS),
                ("new_amount_public_share1", S),
                ("new_in_balance_public_shares1", PMS),
                ("new_out_balance_public_shares1", PMS),
                ("relayer_fee0", FIXED_POINT), ("relayer_fee1", FIXED_POINT),
                ("protocol_fee", FIXED_POINT))),
    11: (_PUB_SETTLE_WITNESS,
         struct(("settlement_obligation", OBLIGATION), ("amount_public_share", S),
                ("in_balance_public_shares", PMS), ("out_balance_public_shares", PMS),
                ("fee_rates", FEE_RATES), ("relayer_fee_recipient", S))),
    12: (_PUB_SETTLE_WITNESS,
         struct(("bounded_match_result", BMR), ("amount_public_share", S),
                ("in_balance_public_shares", PMS), ("out_balance_public_shares", PMS),
                ("internal_relayer_fee", FIXED_POINT),
                ("external_relayer_fee", FIXED_POINT), ("relayer_fee_recipient", S))),
    13: (struct(("intent", INTENT)),
         struct(("settlement_obligation", OBLIGATION), ("relayer_fee", FIXED_POINT),
                ("relayer_fee_recipient", S))),
    14: (struct(("intent", INTENT)),
         struct(("bounded_match_result", BMR), ("internal_relayer_fee", FIXED_POINT),
                ("external_relayer_fee", FIXED_POINT), ("relayer_fee_recipient", S))),
    15: (struct(("note_opening", MERKLE_OPENING)),
         struct(("note", NOTE), ("note_root", S), ("note_nullifier", S))),
    16: (struct(("old_balance", STATE_BALANCE),
                ("old_balance_opening", MERKLE_OPENING)),
         struct(*(_FEE_PUBLIC_STATEMENT_TAIL +
                  [("new_relayer_fee_balance_share", S), ("note", NOTE)]))),
    17: (struct(("old_balance", STATE_BALANCE),
                ("old_balance_opening", MERKLE_OPENING)),
         struct(*(_FEE_PUBLIC_STATEMENT_TAIL +
                  [("new_protocol_fee_balance_share", S), ("note", NOTE)]))),
    18: (struct(("old_balance", STATE_BALANCE),
                ("old_balance_opening", MERKLE_OPENING), ("blinder", S)),
         struct(*(_FEE_PUBLIC_STATEMENT_TAIL +
                  [("new_relayer_fee_balance_share", S),
                   ("relayer_fee_receiver", S), ("note_commitment", S)]))),
    19: (struct(("old_balance", STATE_BALANCE),
                ("old_balance_opening", MERKLE_OPENING), ("blinder", S),
                ("encryption_randomness", S)),
         struct(*(_FEE_PUBLIC_STATEMENT_TAIL +
                  [("new_protocol_fee_balance_share", S),
                   ("protocol_fee_receiver", S), ("note_commitment", S),
                   ("note_ciphertext", ELGAMAL_CT3),
                   ("protocol_encryption_key", JUBJUB_PT)]))),
}


def schema_size(schema):
    """Number of scalar slots a schema flattens to."""
    if schema in (S, U, B):
        return 1
    if schema[0] == "list":
        return schema[2] * schema_size(schema[1])
    return sum(schema_size(sub) for _, sub in schema[1])


def flatten(value, schema, out):
    """Nested JSON value -> flat list of ints (appended to `out`)."""
    if schema == S:
        out.append(int(value))
    elif schema == U:
        out.append(int(value))
    elif schema == B:
        out.append(1 if value else 0)
    elif schema[0] == "list":
        _, sub, n = schema
        if len(value) != n:
            raise ValueError(f"expected list of {n}, got {len(value)}")
        for v in value:
            flatten(v, sub, out)
    else:
        for name, sub in schema[1]:
            if name not in value:
                raise ValueError(f"missing field {name!r}")
            flatten(value[name], sub, out)
    return out


def unflatten(flat, schema, pos=0):
    """Flat scalar list -> (nested JSON value, next position)."""
    if schema == S:
        return str(int(flat[pos])), pos + 1
    if schema == U:
        return int(flat[pos]), pos + 1
    if schema == B:
        return bool(int(flat[pos])), pos + 1
    if schema[0] == "list":
        _, sub, n = schema
        items = []
        for _ in range(n):
            v, pos = unflatten(flat, sub, pos)
            items.append(v)
        return items, pos
    obj = {}
    for name, sub in schema[1]:
        v, pos = unflatten(flat, sub, pos)
        obj[name] = v
    return obj, pos
