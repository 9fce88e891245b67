// PRODUCT PATH — gadget library + darkpool types + circuits (host).
//
// New implementations of the reference's zk-gadget and type layers needed by
// the circuit family (citations inline):
//  - PoseidonHashGadget with fused external/internal sbox+MDS gates
//    (zk_gadgets/primitives/poseidon/hash.rs:69-423, gates.rs:73-179)
//  - CSPRNG / stream-cipher / commitment / recovery-id gadgets
//    (zk_gadgets/state_primitives/{csprng,stream_cipher,commitment,
//     recovery_id}.rs)
//  - PoseidonCSPRNG + StateWrapper native semantics
//    (darkpool-types/src/{csprng.rs,state_wrapper.rs})
//  - darkpool value types' scalar layouts (darkpool-types/src/{balance.rs,
//    deposit.rs}; circuit-types/src/primitives/schnorr.rs)
//  - the `Valid Balance Create` circuit
//    (zk_circuits/valid_balance_create.rs:44-133) and its fixed-seed
//    witness/statement builder mirroring test_helpers (:225-283).
#pragma once
#include <vector>
#include "jubjub.hpp"
#include "plonk_circuit.hpp"
#include "poseidon2.hpp"
#include "test_circuits.hpp"  // Lcg

namespace rng {

// =============================== gadgets ===============================

// In-circuit Poseidon2 sponge with fused gates (hash.rs:69-423).
struct PoseidonHashGadget {
    std::array<Var, 3> state;
    int next_index = 0;
    bool squeezing = false;

    explicit PoseidonHashGadget(PlonkCircuit& cs) {
        state = {cs.zero(), cs.zero(), cs.zero()};
    }
    void reset(PlonkCircuit& cs) {
        state = {cs.zero(), cs.zero(), cs.zero()};
        next_index = 0;
        squeezing = false;
    }

    // fused external sbox+MDS gate for one output element (gates.rs:73-101):
    // out = rc + sbox?(curr) + sbox?(s0) + sbox?(s1) + sbox?(s2)
    static Var fused_external(PlonkCircuit& cs, bool sbox, const Fr& rc, Var curr,
                              const std::array<Var, 3>& s) {
        auto p5v = [&](Var x) { return Poseidon2::sbox(cs.witness(x)); };
        Fr out = rc;
        if (sbox) {
            out = out.add(p5v(curr)).add(p5v(s[0])).add(p5v(s[1])).add(p5v(s[2]));
        } else {
            out = out.add(cs.witness(curr)).add(cs.witness(s[0])).add(cs.witness(s[1]))
                      .add(cs.witness(s[2]));
        }
        Var o = cs.create_variable(out);
        std::array<Fr, NUM_SELECTORS> q{};
        for (auto& x : q) x = Fr::zero();
        if (sbox) {
            q[SEL_HASH0] = q[SEL_HASH1] = q[SEL_HASH2] = q[SEL_HASH3] = Fr::one();
        } else {
            q[SEL_LC0] = q[SEL_LC1] = q[SEL_LC2] = q[SEL_LC3] = Fr::one();
        }
        q[SEL_C] = rc;
        q[SEL_O] = Fr::one();
        cs.insert_gate({curr, s[0], s[1], s[2], o}, q);
        return o;
    }

    // fused internal sbox+MDS gate (gates.rs:119-180):
    // out = rc + coeff*sbox?(curr) + pow5(s0) + s1 + s2
    static Var fused_internal(PlonkCircuit& cs, bool sbox, const Fr& rc, const Fr& coeff,
                              Var curr, const std::array<Var, 3>& s) {
        Fr cval = cs.witness(curr);
        Fr celem = sbox ? Poseidon2::sbox(cval) : cval;
        Fr out = rc.add(coeff.mul(celem)).add(Poseidon2::sbox(cs.witness(s[0])))
                     .add(cs.witness(s[1])).add(cs.witness(s[2]));
        Var o = cs.create_variable(out);
        std::array<Fr, NUM_SELECTORS> q{};
        for (auto& x : q) x = Fr::zero();
        if (sbox) {
            q[SEL_HASH0] = coeff;
            q[SEL_HASH1] = Fr::one();
            q[SEL_LC2] = q[SEL_LC3] = Fr::one();
        } else {
            q[SEL_HASH1] = Fr::one();
            q[SEL_LC0] = coeff;
            q[SEL_LC2] = q[SEL_LC3] = Fr::one();
        }
        q[SEL_C] = rc;
        q[SEL_O] = Fr::one();
        cs.insert_gate({curr, s[0], s[1], s[2], o}, q);
        return o;
    }

    void external_round(PlonkCircuit& cs, bool sbox, const Fr rc[3]) {
        std::array<Var, 3> in = state;
        for (int i = 0; i < 3; ++i)
            state[i] = fused_external(cs, sbox, rc[i], in[i], in);
    }
    void internal_round(PlonkCircuit& cs, const Fr rc[3]) {
        std::array<Var, 3> in = state;
        state[0] = fused_internal(cs, true, rc[0], Fr::one(), in[0], in);
        state[1] = fused_internal(cs, false, rc[1], Fr::one(), in[1], in);
        state[2] = fused_internal(cs, false, rc[2], Fr::from_u64(2), in[2], in);
    }

    // permutation with inter-round constant fusion (hash.rs:202-252)
    void permute(PlonkCircuit& cs) {
        constexpr int HALF = Poseidon2::R_F / 2;
        Fr rc[3];
        for (int w = 0; w < 3; ++w) rc[w] = Poseidon2::rc_full(0, w);
        external_round(cs, false, rc);  // initial MDS + round-0 constants
        for (int r = 0; r < HALF - 1; ++r) {
            for (int w = 0; w < 3; ++w) rc[w] = Poseidon2::rc_full(r + 1, w);
            external_round(cs, true, rc);
        }
        rc[0] = Poseidon2::rc_partial(0);
        rc[1] = rc[2] = Fr::zero();
        external_round(cs, true, rc);  // last ext round fused w/ 1st partial rc
        for (int r = 0; r < Poseidon2::R_P - 1; ++r) {
            rc[0] = Poseidon2::rc_partial(r + 1);
            rc[1] = rc[2] = Fr::zero();
            internal_round(cs, rc);
        }
        for (int w = 0; w < 3; ++w) rc[w] = Poseidon2::rc_full(HALF, w);
        internal_round(cs, rc);  // last internal fused w/ next external rc
        for (int r = HALF; r < Poseidon2::R_F - 1; ++r) {
            for (int w = 0; w < 3; ++w) rc[w] = Poseidon2::rc_full(r + 1, w);
            external_round(cs, true, rc);
        }
        rc[0] = rc[1] = rc[2] = Fr::zero();
        external_round(cs, true, rc);  // final round, no trailing constant
    }

    void absorb(PlonkCircuit& cs, Var a) {
        if (next_index == Poseidon2::RATE) {
            permute(cs);
            next_index = 0;
        }
        int at = next_index + Poseidon2::CAPACITY;
        state[at] = cs.add(a, state[at]);
        next_index++;
    }
    void batch_absorb(PlonkCircuit& cs, const std::vector<Var>& vs) {
        for (Var v : vs) absorb(cs, v);
    }
    Var squeeze(PlonkCircuit& cs) {
        if (!squeezing || next_index == Poseidon2::RATE) {
            permute(cs);
            next_index = 0;
            squeezing = true;
        }
        return state[Poseidon2::CAPACITY + next_index++];
    }
    Var hash(PlonkCircuit& cs, const std::vector<Var>& input) {
        batch_absorb(cs, input);
        return squeeze(cs);
    }
};

// CSPRNG state variable (darkpool-types/src/csprng.rs: {seed, index})
struct CsprngVar {
    Var seed, index;
};

// CSPRNGGadget (state_primitives/csprng.rs): next = H(seed, index); index++
inline Var csprng_next(PlonkCircuit& cs, CsprngVar& st) {
    PoseidonHashGadget h(cs);
    Var out = h.hash(cs, {st.seed, st.index});
    st.index = cs.add(st.index, cs.one());
    return out;
}
inline std::vector<Var> csprng_next_k(PlonkCircuit& cs, CsprngVar& st, size_t k) {
    std::vector<Var> out;
    for (size_t i = 0; i < k; ++i) out.push_back(csprng_next(cs, st));
    return out;
}

// StreamCipherGadget::encrypt (state_primitives/stream_cipher.rs:22-39):
// pads = next_k, ciphertext_i = value_i - pad_i; returns (private, public)
inline void stream_cipher_encrypt(PlonkCircuit& cs, const std::vector<Var>& values,
                                  CsprngVar& st, std::vector<Var>& private_share,
                                  std::vector<Var>& public_share) {
    private_share = csprng_next_k(cs, st, values.size());
    public_share.clear();
    for (size_t i = 0; i < values.size(); ++i)
        public_share.push_back(cs.sub(values[i], private_share[i]));
}

// resumable commitment (commitment.rs:433-446): comm = v0; comm = H(comm, v_i)
inline Var resumable_commitment(PlonkCircuit& cs, const std::vector<Var>& vs) {
    PoseidonHashGadget h(cs);
    Var comm = vs[0];
    for (size_t i = 1; i < vs.size(); ++i) {
        comm = h.hash(cs, {comm, vs[i]});
        h.reset(cs);
    }
    return comm;
}

// CommitmentGadget::compute_commitment (commitment.rs:72-135):
//   private = H(private_shares || recovery_stream || share_stream)
//   public  = resumable(public_shares)
//   comm    = H(private, public)
inline Var commitment_gadget(PlonkCircuit& cs, const std::vector<Var>& private_share,
                             const CsprngVar& recovery, const CsprngVar& share,
                             const std::vector<Var>& public_share) {
    PoseidonHashGadget h(cs);
    std::vector<Var> inputs = private_share;
    inputs.push_back(recovery.seed);
    inputs.push_back(recovery.index);
    inputs.push_back(share.seed);
    inputs.push_back(share.index);
    Var priv = h.hash(cs, inputs);
    Var pub = resumable_commitment(cs, public_share);
    PoseidonHashGadget h2(cs);
    return h2.hash(cs, {priv, pub});
}

// ============================ native types ============================

// PoseidonCSPRNG (csprng.rs): next = H(seed, index); index++
struct Csprng {
    Fr seed;
    uint64_t index = 0;
    Fr next() {
        Fr in[2] = {seed, Fr::from_u64(index)};
        index++;
        return poseidon_hash(in, 2);
    }
};

struct Balance {  // DarkpoolBalance (balance.rs): 8 scalars in field order
    Fr mint, owner, relayer_fee_recipient, authority_x, authority_y,
        relayer_fee_balance, protocol_fee_balance, amount;
    std::vector<Fr> to_scalars() const {
        return {mint, owner, relayer_fee_recipient, authority_x, authority_y,
                relayer_fee_balance, protocol_fee_balance, amount};
    }
    void from_scalars(const Fr* s) {
        mint = s[0];
        owner = s[1];
        relayer_fee_recipient = s[2];
        authority_x = s[3];
        authority_y = s[4];
        relayer_fee_balance = s[5];
        protocol_fee_balance = s[6];
        amount = s[7];
    }
    static constexpr size_t NUM_SCALARS = 8;
};

struct Deposit {  // deposit.rs: 3 scalars
    Fr from, token, amount;
    std::vector<Fr> to_scalars() const { return {from, token, amount}; }
};

// native commitment over a state-wrapped element (state_wrapper.rs:125-163)
inline Fr native_commitment(const std::vector<Fr>& private_shares,
                            const Csprng& recovery, const Csprng& share,
                            const std::vector<Fr>& public_shares) {
    std::vector<Fr> in = private_shares;
    in.push_back(recovery.seed);
    in.push_back(Fr::from_u64(recovery.index));
    in.push_back(share.seed);
    in.push_back(Fr::from_u64(share.index));
    Fr priv = poseidon_hash(in.data(), in.size());
    Fr pub = public_shares[0];
    for (size_t i = 1; i < public_shares.size(); ++i) {
        Fr two[2] = {pub, public_shares[i]};
        pub = poseidon_hash(two, 2);
    }
    Fr fin[2] = {priv, pub};
    return poseidon_hash(fin, 2);
}

// ====================== Valid Balance Create ======================

struct VbcWitness {  // valid_balance_create.rs:143-150 (12 scalars)
    Csprng initial_share_stream;
    Csprng initial_recovery_stream;
    Balance balance;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = {initial_share_stream.seed, Fr::from_u64(initial_share_stream.index),
                             initial_recovery_stream.seed,
                             Fr::from_u64(initial_recovery_stream.index)};
        auto b = balance.to_scalars();
        v.insert(v.end(), b.begin(), b.end());
        return v;
    }
};

struct VbcStatement {  // valid_balance_create.rs:159-170 (13 scalars)
    Deposit deposit;
    Fr balance_commitment;
    Fr recovery_id;
    std::vector<Fr> new_balance_share;  // 8
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = deposit.to_scalars();
        v.push_back(balance_commitment);
        v.push_back(recovery_id);
        v.insert(v.end(), new_balance_share.begin(), new_balance_share.end());
        return v;
    }
};

// fixed-seed witness/statement builder mirroring test_helpers
// (valid_balance_create.rs:225-283 + darkpool-types/src/fuzzing.rs:58-71,
//  324-344): addresses are 160-bit, amounts < 2^100 (AMOUNT_BITS=100,
//  circuit-types/src/lib.rs:59), stream states random with random indices.
inline void vbc_build_witness_statement(uint64_t seed, VbcWitness& w, VbcStatement& st) {
    Lcg rng(seed);
    auto addr = [&]() {  // 160-bit address scalar (fuzzing.rs:66-71)
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    Fr amount = Fr::from_u64(rng.next());  // < 2^53 < 2^100

    st.deposit = {addr(), addr(), amount};
    w.balance = {st.deposit.token, st.deposit.from, addr(), rng.fr(), rng.fr(),
                 Fr::zero(), Fr::zero(), amount};
    w.initial_share_stream = {rng.fr(), rng.next()};
    w.initial_recovery_stream = {rng.fr(), rng.next()};

    // encrypt the balance: pads from the share stream, public = value - pad
    Csprng share = w.initial_share_stream;
    auto values = w.balance.to_scalars();
    std::vector<Fr> pads, pub;
    for (auto& v : values) {
        Fr p = share.next();
        pads.push_back(p);
        pub.push_back(v.sub(p));
    }
    st.new_balance_share = pub;

    // recovery id + commitment over the UPDATED streams
    Csprng recovery = w.initial_recovery_stream;
    st.recovery_id = recovery.next();
    st.balance_commitment = native_commitment(pads, recovery, share, pub);
}

// scalar-vector (de)serialization, inverse of to_scalars (BaseType
// from_scalars semantics, traits.rs:103-118)
inline void vbc_witness_from_scalars(const Fr* s, VbcWitness& w) {
    u64 c[4];
    w.initial_share_stream.seed = s[0];
    s[1].to_canonical(c);
    w.initial_share_stream.index = c[0];
    w.initial_recovery_stream.seed = s[2];
    s[3].to_canonical(c);
    w.initial_recovery_stream.index = c[0];
    w.balance.from_scalars(s + 4);
}
inline void vbc_statement_from_scalars(const Fr* s, VbcStatement& st) {
    st.deposit = {s[0], s[1], s[2]};
    st.balance_commitment = s[3];
    st.recovery_id = s[4];
    st.new_balance_share.assign(s + 5, s + 13);
}
// apply_constraints (valid_balance_create.rs:44-133).  Returns nothing; all
// witness/statement vars are allocated in order (create_witness then
// create_public_var, traits.rs:984-991).
inline void vbc_apply_constraints(PlonkCircuit& cs, const VbcWitness& w,
                                  const VbcStatement& st) {
    // --- allocate witness vars (field order) ---
    CsprngVar share{cs.create_variable(w.initial_share_stream.seed),
                    cs.create_variable(Fr::from_u64(w.initial_share_stream.index))};
    CsprngVar recovery{cs.create_variable(w.initial_recovery_stream.seed),
                       cs.create_variable(Fr::from_u64(w.initial_recovery_stream.index))};
    auto bscal = w.balance.to_scalars();
    std::vector<Var> balance;
    for (auto& s : bscal) balance.push_back(cs.create_variable(s));

    // --- allocate statement vars as public inputs ---
    auto sscal = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : sscal) pub.push_back(cs.create_public_variable(s));
    Var dep_from = pub[0], dep_token = pub[1], dep_amount = pub[2];
    Var st_commitment = pub[3], st_recovery_id = pub[4];
    std::vector<Var> st_share(pub.begin() + 5, pub.begin() + 13);

    // 1. validate deposit (:71-87)
    cs.enforce_in_range(dep_amount, 100);  // AmountGadget, AMOUNT_BITS=100
    cs.enforce_equal(dep_token, balance[0]);  // token == mint
    cs.enforce_equal(dep_from, balance[1]);   // from == owner
    // 2. validate new balance (:90-104)
    cs.enforce_equal(balance[7], dep_amount);  // amount
    cs.enforce_equal(balance[5], cs.zero());   // relayer fee
    cs.enforce_equal(balance[6], cs.zero());   // protocol fee
    // 3. encryption (:110-133)
    std::vector<Var> private_share, public_share;
    stream_cipher_encrypt(cs, balance, share, private_share, public_share);
    for (int i = 0; i < 8; ++i) cs.enforce_equal(public_share[i], st_share[i]);
    // 4. recovery id (:58-59)
    Var rid = csprng_next(cs, recovery);
    cs.enforce_equal(rid, st_recovery_id);
    // 5. commitment (:64-65) — uses the UPDATED stream states
    Var comm = commitment_gadget(cs, private_share, recovery, share, public_share);
    cs.enforce_equal(comm, st_commitment);
}

// ---- equality / zero gadgets (comparators.rs EqGadget/NotEqualGadget) ----
// z = (a == 0): allocate z (bool) and iw (= a^-1 or 0); constrain
//   a*iw + z - 1 = 0   and   a*z = 0
inline Var is_zero_gadget(PlonkCircuit& cs, Var a) {
    Fr av = cs.witness(a);
    bool zero = av.is_zero();
    Var z = cs.create_boolean_variable(zero ? Fr::one() : Fr::zero());
    Var iw = cs.create_variable(zero ? Fr::zero() : av.inverse());
    {
        std::array<Fr, NUM_SELECTORS> q{};
        for (auto& x : q) x = Fr::zero();
        q[SEL_MUL0] = Fr::one();
        q[SEL_LC2] = Fr::one();
        q[SEL_C] = Fr::one().neg();
        cs.insert_gate({a, iw, z, 0, 0}, q);
    }
    {
        std::array<Fr, NUM_SELECTORS> q{};
        for (auto& x : q) x = Fr::zero();
        q[SEL_MUL0] = Fr::one();
        cs.insert_gate({a, z, 0, 0, 0}, q);
    }
    return z;
}
inline void constrain_not_equal(PlonkCircuit& cs, Var a, Var b) {
    Var d = cs.sub(a, b);
    Var z = is_zero_gadget(cs, d);
    cs.enforce_false(z);
}

// ---- state wrapper vars + nullifier/merkle gadgets ----
struct StateWrapperVars {  // StateWrapperVar (state_wrapper.rs)
    CsprngVar recovery, share;
    std::vector<Var> inner, public_share;  // 8 each for a balance
};

// NullifierGadget::compute_nullifier (state_primitives/nullifier.rs:21-40)
inline Var nullifier_gadget(PlonkCircuit& cs, const StateWrapperVars& el) {
    constrain_not_equal(cs, el.recovery.index, cs.zero());
    Var last_idx = cs.sub(el.recovery.index, cs.one());
    PoseidonHashGadget h(cs);
    Var rid = h.hash(cs, {el.recovery.seed, last_idx});  // CSPRNGGadget::get_ith
    PoseidonHashGadget h2(cs);
    return h2.hash(cs, {rid, el.recovery.seed});
}

// PoseidonMerkleHashGadget::compute_root_prehashed (merkle.rs:29-95)
inline Var merkle_root_gadget(PlonkCircuit& cs, Var leaf, const std::vector<Var>& elems,
                              const std::vector<Var>& indices /*bools; true=right child*/) {
    Var cur = leaf;
    for (size_t i = 0; i < elems.size(); ++i) {
        Var left = cs.mux(indices[i], elems[i], cur);
        std::array<Fr, 4> cf{Fr::one(), Fr::one(), Fr::one().neg(), Fr::zero()};
        Var right = cs.lc({cur, elems[i], left, cs.zero()}, cf);
        PoseidonHashGadget h(cs);
        cur = h.hash(cs, {left, right});
    }
    return cur;
}

// natives
inline Fr native_nullifier(const Csprng& recovery) {
    Csprng c = recovery;
    c.index -= 1;
    Fr rid = c.next();
    Fr in[2] = {rid, recovery.seed};
    return poseidon_hash(in, 2);
}
inline Fr native_merkle_root(const Fr& leaf, const std::vector<Fr>& elems,
                             const std::vector<bool>& indices) {
    Fr cur = leaf;
    for (size_t i = 0; i < elems.size(); ++i) {
        Fr l = indices[i] ? elems[i] : cur;
        Fr r = indices[i] ? cur : elems[i];
        Fr in[2] = {l, r};
        cur = poseidon_hash(in, 2);
    }
    return cur;
}

// ================ embedded-curve gadgets (Baby Jubjub) ================
// In-circuit twisted Edwards arithmetic over the embedded curve
// (mpc-relation gadgets/ecc + jf-primitives circuit/signature/schnorr.rs,
// circuit/elgamal.rs shapes; see jubjub.hpp header for the hash-choice
// parity note).  Addition is COMPLETE (a QR, d non-QR — verified by
// scripts/gen_babyjubjub_params.py), so identity needs no special casing.

struct JjPointVars {
    Var x, y;
};

inline JjPointVars jj_const_point_gadget(PlonkCircuit& cs, const JjPoint& p) {
    Var x = cs.create_variable(p.x);
    cs.enforce_constant(x, p.x);
    Var y = cs.create_variable(p.y);
    cs.enforce_constant(y, p.y);
    return {x, y};
}

// constrain a*x^2 + y^2 == 1 + d*x^2*y^2
inline void jj_on_curve_gadget(PlonkCircuit& cs, const JjPointVars& p) {
    Var x2 = cs.mul(p.x, p.x);
    Var y2 = cs.mul(p.y, p.y);
    Var x2y2 = cs.mul(x2, y2);
    std::array<Fr, NUM_SELECTORS> q{};
    q[SEL_LC0] = jj_a();
    q[SEL_LC1] = Fr::one();
    q[SEL_LC2] = jj_d().neg();
    q[SEL_C] = Fr::one().neg();
    cs.insert_gate({x2, y2, x2y2, 0, 0}, q);
}

// complete addition: 5 mul gates + 2 rational-constraint gates
inline JjPointVars jj_add_gadget(PlonkCircuit& cs, const JjPointVars& p,
                                 const JjPointVars& q_) {
    Var x1x2 = cs.mul(p.x, q_.x);
    Var y1y2 = cs.mul(p.y, q_.y);
    Var x1y2 = cs.mul(p.x, q_.y);
    Var y1x2 = cs.mul(p.y, q_.x);
    Var tau = cs.mul(x1x2, y1y2);
    Fr t = jj_d().mul(cs.witness(tau));
    Fr x3v = cs.witness(x1y2).add(cs.witness(y1x2)).mul(Fr::one().add(t).inverse());
    Fr y3v = cs.witness(y1y2)
                 .sub(jj_a().mul(cs.witness(x1x2)))
                 .mul(Fr::one().sub(t).inverse());
    Var x3 = cs.create_variable(x3v);
    Var y3 = cs.create_variable(y3v);
    {  // x3 + d*x3*tau - x1y2 - y1x2 = 0
        std::array<Fr, NUM_SELECTORS> q{};
        q[SEL_LC0] = Fr::one();
        q[SEL_MUL0] = jj_d();
        q[SEL_LC2] = Fr::one().neg();
        q[SEL_LC3] = Fr::one().neg();
        cs.insert_gate({x3, tau, x1y2, y1x2, 0}, q);
    }
    {  // y3 - d*y3*tau - y1y2 + a*x1x2 = 0
        std::array<Fr, NUM_SELECTORS> q{};
        q[SEL_LC0] = Fr::one();
        q[SEL_MUL0] = jj_d().neg();
        q[SEL_LC2] = Fr::one().neg();
        q[SEL_LC3] = jj_a();
        cs.insert_gate({y3, tau, y1y2, x1x2, 0}, q);
    }
    return {x3, y3};
}

inline JjPointVars jj_select_gadget(PlonkCircuit& cs, Var b, const JjPointVars& t,
                                    const JjPointVars& f) {
    return {cs.mux(b, t.x, f.x), cs.mux(b, t.y, f.y)};
}

// bits little-endian (boolean vars); double-and-add from the LSB
inline JjPointVars jj_scalar_mul_gadget(PlonkCircuit& cs, const std::vector<Var>& bits,
                                        JjPointVars base) {
    JjPointVars acc{cs.zero(), cs.one()};
    JjPointVars addend = base;
    for (size_t i = 0; i < bits.size(); ++i) {
        JjPointVars sum = jj_add_gadget(cs, acc, addend);
        acc = jj_select_gadget(cs, bits[i], sum, acc);
        if (i + 1 < bits.size()) addend = jj_add_gadget(cs, addend, addend);
    }
    return acc;
}

// Schnorr verification (schnorr.rs:43-54 shape): s*B == R + c*V with
// c = low 248 bits of Poseidon2(vk || R || msg).  `s` is the plain scalar
// witnessed as a field element (< 2^251).
inline void schnorr_verify_gadget(PlonkCircuit& cs, const JjPointVars& vk,
                                  const JjPointVars& R, Var s,
                                  const std::vector<Var>& msg) {
    jj_on_curve_gadget(cs, vk);
    jj_on_curve_gadget(cs, R);
    PoseidonHashGadget h(cs);
    std::vector<Var> in = {vk.x, vk.y, R.x, R.y};
    in.insert(in.end(), msg.begin(), msg.end());
    Var c = h.hash(cs, in);
    auto cbits = cs.to_bits(c, 254);
    cbits.resize(248);  // challenge scalar = low 248 bits (< l)
    auto sbits = cs.to_bits(s, JJ_ORDER_BITS);
    JjPointVars B = jj_const_point_gadget(cs, jj_base());
    JjPointVars sB = jj_scalar_mul_gadget(cs, sbits, B);
    JjPointVars cV = jj_scalar_mul_gadget(cs, cbits, vk);
    JjPointVars rhs = jj_add_gadget(cs, R, cV);
    cs.enforce_equal(sB.x, rhs.x);
    cs.enforce_equal(sB.y, rhs.y);
}

// ElGamal hybrid encryption (elgamal.rs shape): eph = k*B, shared = k*pk,
// pad_i = Poseidon2(shared.x, shared.y, i), c_i = m_i + pad_i.  Returns the
// ephemeral key and ciphertext vars; `k` witnessed as a field element.
inline void elgamal_encrypt_gadget(PlonkCircuit& cs, const JjPointVars& pk, Var k,
                                   const std::vector<Var>& msg, JjPointVars& out_eph,
                                   std::vector<Var>& out_cipher) {
    jj_on_curve_gadget(cs, pk);
    auto kbits = cs.to_bits(k, JJ_ORDER_BITS);
    JjPointVars B = jj_const_point_gadget(cs, jj_base());
    out_eph = jj_scalar_mul_gadget(cs, kbits, B);
    JjPointVars shared = jj_scalar_mul_gadget(cs, kbits, pk);
    out_cipher.clear();
    for (size_t i = 0; i < msg.size(); ++i) {
        PoseidonHashGadget h(cs);
        Var idx = cs.create_variable(Fr::from_u64((u64)i));
        cs.enforce_constant(idx, Fr::from_u64((u64)i));
        Var pad = h.hash(cs, {shared.x, shared.y, idx});
        out_cipher.push_back(cs.add(msg[i], pad));
    }
}

// ================== Intent And Balance Private Settlement ==================
// (zk_circuits/settlement/intent_and_balance_private_settlement.rs — the
//  VALID MATCH MPC successor; BASELINE config #4)

constexpr int AMOUNT_BITS = 100;       // circuit-types/src/lib.rs:59
constexpr int FP_PRECISION = 63;       // fixed_point.rs:41 (repr = x * 2^63)

// GreaterThanEqGadget (comparators.rs:259-267): range-check a-b in `bits`
inline void gte_gadget(PlonkCircuit& cs, Var a, Var b, int bits) {
    Var diff = cs.sub(a, b);
    cs.enforce_in_range(diff, bits);
}

// FixedPointGadget::floor (fixed_point.rs gadget:87-133):
// allocate floor = repr >> 63 and constrain repr - 2^63*floor in [0, 2^63)
inline Var fp_floor_gadget(PlonkCircuit& cs, Var repr) {
    u64 limbs[4];
    cs.witness(repr).to_canonical(limbs);
    // shift right by 63 (value < 2^227 in valid uses)
    u64 fl[4];
    for (int i = 0; i < 4; ++i) {
        u64 lo = limbs[i] >> 63;
        u64 hi = (i + 1 < 4) ? (limbs[i + 1] << 1) : 0;
        fl[i] = lo | hi;
    }
    Var floor_v = cs.create_variable(Fr::from_canonical(fl));
    Fr two63 = Fr::from_u64(1ull << 62).dbl();
    std::array<Fr, 4> cf{Fr::one(), two63.neg(), Fr::zero(), Fr::zero()};
    Var diff = cs.lc({repr, floor_v, cs.zero(), cs.zero()}, cf);
    cs.enforce_in_range(diff, FP_PRECISION);
    return floor_v;
}

struct FeeTakeVars {
    Var relayer_fee, protocol_fee;
};

// FeeGadget::compute_fee_take (state_gadgets/fee.rs:21-37)
inline FeeTakeVars fee_take_gadget(PlonkCircuit& cs, Var amount, Var relayer_rate_repr,
                                   Var protocol_rate_repr) {
    Var rf_fp = cs.mul(relayer_rate_repr, amount);   // mul_integer
    Var pf_fp = cs.mul(protocol_rate_repr, amount);
    return {fp_floor_gadget(cs, rf_fp), fp_floor_gadget(cs, pf_fp)};
}

struct Obligation {  // settlement_obligation.rs:36-47 (4 scalars)
    Fr input_token, output_token, amount_in, amount_out;
    std::vector<Fr> to_scalars() const {
        return {input_token, output_token, amount_in, amount_out};
    }
};
struct Intent {  // intent.rs:49-70 (5 scalars; min_price = FixedPoint repr)
    Fr in_token, out_token, owner, min_price_repr, amount_in;
    std::vector<Fr> to_scalars() const {
        return {in_token, out_token, owner, min_price_repr, amount_in};
    }
};
struct PostMatchShare {  // balance.rs:145-152 PostMatchBalance(Share) (3 scalars)
    // Reference field order: relayer_fee_balance, protocol_fee_balance, amount.
    Fr relayer_fee_balance, protocol_fee_balance, amount;
    std::vector<Fr> to_scalars() const {
        return {relayer_fee_balance, protocol_fee_balance, amount};
    }
};

struct SettlementParty {
    Obligation obligation;
    Intent intent;
    Fr pre_amount_share;
    Balance input_balance;
    PostMatchShare pre_in_shares;
    Balance output_balance;
    PostMatchShare pre_out_shares;
};

struct SettlementWitness {  // 2 * 32 = 64 scalars, field order per the struct
    SettlementParty p[2];
};
struct SettlementStatement {  // 17 scalars
    Fr new_amount_share[2];
    PostMatchShare new_in_shares[2];
    PostMatchShare new_out_shares[2];
    Fr relayer_fee_repr[2];
    Fr protocol_fee_repr;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v;
        for (int i = 0; i < 2; ++i) {
            v.push_back(new_amount_share[i]);
            auto a = new_in_shares[i].to_scalars();
            v.insert(v.end(), a.begin(), a.end());
            auto b = new_out_shares[i].to_scalars();
            v.insert(v.end(), b.begin(), b.end());
        }
        v.push_back(relayer_fee_repr[0]);
        v.push_back(relayer_fee_repr[1]);
        v.push_back(protocol_fee_repr);
        return v;
    }
};

inline void settlement_witness_from_scalars(const Fr* s, SettlementWitness& w) {
    for (int i = 0; i < 2; ++i) {
        const Fr* p = s + 32 * i;
        w.p[i].obligation = {p[0], p[1], p[2], p[3]};
        w.p[i].intent = {p[4], p[5], p[6], p[7], p[8]};
        w.p[i].pre_amount_share = p[9];
        w.p[i].input_balance.from_scalars(p + 10);
        w.p[i].pre_in_shares = {p[18], p[19], p[20]};
        w.p[i].output_balance.from_scalars(p + 21);
        w.p[i].pre_out_shares = {p[29], p[30], p[31]};
    }
}
inline void settlement_statement_from_scalars(const Fr* s, SettlementStatement& st) {
    for (int i = 0; i < 2; ++i) {
        const Fr* p = s + 7 * i;
        st.new_amount_share[i] = p[0];
        st.new_in_shares[i] = {p[1], p[2], p[3]};
        st.new_out_shares[i] = {p[4], p[5], p[6]};
    }
    st.relayer_fee_repr[0] = s[14];
    st.relayer_fee_repr[1] = s[15];
    st.protocol_fee_repr = s[16];
}

// native statement share updates from the witness pre-update shares
// (test_helpers :356-390): recomputable after the pre shares are replaced by
// validity-circuit outputs in a bundle.
inline void settlement_update_statement(const SettlementWitness& w,
                                        SettlementStatement& st) {
    auto floor_mul = [](const Fr& rate_repr, const Fr& amt) {
        // floor(rate * amt / 2^63) over canonical integers (fits 128 bits here)
        u64 rl[4], al[4];
        rate_repr.to_canonical(rl);
        amt.to_canonical(al);
        unsigned __int128 rate = ((unsigned __int128)rl[1] << 64) | rl[0];
        unsigned __int128 prod = rate * al[0];  // amounts < 2^64 here
        unsigned __int128 fl = prod >> 63;
        u64 l[4] = {(u64)fl, (u64)(fl >> 64), 0, 0};
        return Fr::from_canonical(l);
    };
    for (int i = 0; i < 2; ++i) {
        const auto& ob = w.p[i].obligation;
        Fr rf = floor_mul(st.relayer_fee_repr[i], ob.amount_out);
        Fr pf = floor_mul(st.protocol_fee_repr, ob.amount_out);
        Fr net = ob.amount_out.sub(rf).sub(pf);
        st.new_amount_share[i] = w.p[i].pre_amount_share.sub(ob.amount_in);
        st.new_in_shares[i] = {w.p[i].pre_in_shares.relayer_fee_balance,
                               w.p[i].pre_in_shares.protocol_fee_balance,
                               w.p[i].pre_in_shares.amount.sub(ob.amount_in)};
        st.new_out_shares[i] = {w.p[i].pre_out_shares.relayer_fee_balance.add(rf),
                                w.p[i].pre_out_shares.protocol_fee_balance.add(pf),
                                w.p[i].pre_out_shares.amount.add(net)};
    }
}

// fixed-seed witness/statement (test_helpers create_witness_statement,
// intent_and_balance_private_settlement.rs:334-420; f64 price sampling
// replaced by exact integer fixed-point construction)
inline void settlement_build_witness_statement(uint64_t seed, SettlementWitness& w,
                                               SettlementStatement& st) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    auto amount_u64 = [&]() { return (rng.next() & ((1ull << 52) - 1)) + 2; };
    Fr token0 = addr(), token1 = addr();
    uint64_t t0_amount = amount_u64(), t1_amount = amount_u64();
    uint64_t t0_traded = t0_amount / 2 + 1, t1_traded = t1_amount / 2 + 1;

    // min_price = half the trade price, as exact 2^63 fixed point:
    // repr = floor((traded_out << 63) / traded_in / 2)
    auto half_price_repr = [&](uint64_t out_amt, uint64_t in_amt) {
        unsigned __int128 r = ((unsigned __int128)out_amt << 63) / in_amt / 2;
        u64 l[4] = {(u64)r, (u64)(r >> 64), 0, 0};
        return Fr::from_canonical(l);
    };

    Obligation ob0{token0, token1, Fr::from_u64(t0_traded), Fr::from_u64(t1_traded)};
    Obligation ob1{token1, token0, Fr::from_u64(t1_traded), Fr::from_u64(t0_traded)};
    Fr owner0 = addr(), owner1 = addr();
    Intent in0{token0, token1, owner0, half_price_repr(t1_traded, t0_traded),
               Fr::from_u64(t0_amount)};
    Intent in1{token1, token0, owner1, half_price_repr(t0_traded, t1_traded),
               Fr::from_u64(t1_amount)};

    auto send_balance = [&](const Fr& owner, const Obligation& ob, uint64_t amt) {
        return Balance{ob.input_token, owner, addr(), rng.fr(), rng.fr(),
                       Fr::from_u64(rng.next() & 0xFFFF), Fr::from_u64(rng.next() & 0xFFFF),
                       Fr::from_u64(amt)};
    };
    auto recv_balance = [&](const Fr& owner, const Obligation& ob) {
        return Balance{ob.output_token, owner, addr(), rng.fr(), rng.fr(),
                       Fr::from_u64(rng.next() & 0xFFFF), Fr::from_u64(rng.next() & 0xFFFF),
                       Fr::from_u64(rng.next() & ((1ull << 40) - 1))};
    };
    w.p[0] = {ob0, in0, rng.fr(), send_balance(owner0, ob0, t0_amount), {rng.fr(), rng.fr(), rng.fr()},
              recv_balance(owner0, ob0), {rng.fr(), rng.fr(), rng.fr()}};
    w.p[1] = {ob1, in1, rng.fr(), send_balance(owner1, ob1, t1_amount), {rng.fr(), rng.fr(), rng.fr()},
              recv_balance(owner1, ob1), {rng.fr(), rng.fr(), rng.fr()}};

    // fee rates: ~0.1% as exact fixed point
    auto fee_repr = [&]() {
        unsigned __int128 r = ((unsigned __int128)1 << 63) / (1000 + (rng.next() & 1023));
        u64 l[4] = {(u64)r, (u64)(r >> 64), 0, 0};
        return Fr::from_canonical(l);
    };
    st.relayer_fee_repr[0] = fee_repr();
    st.relayer_fee_repr[1] = fee_repr();
    st.protocol_fee_repr = fee_repr();

    settlement_update_statement(w, st);
}

struct SettlementVars {
    std::array<Var, 4> ob;
    std::array<Var, 5> intent;
    Var pre_amount;
    std::array<Var, 8> in_bal;
    std::array<Var, 3> pre_in;
    std::array<Var, 8> out_bal;
    std::array<Var, 3> pre_out;
};

// apply_constraints (intent_and_balance_private_settlement.rs:46-158 +
// settlement_lib.rs:30-199).  Creates the four proof-linking groups.
inline void settlement_apply_constraints(PlonkCircuit& cs, const SettlementWitness& w,
                                         const SettlementStatement& st) {
    const char* party_group[2] = {"intent_and_balance_settlement_party0",
                                  "intent_and_balance_settlement_party1"};
    const char* out_group[2] = {"output_balance_settlement_party0",
                                "output_balance_settlement_party1"};
    for (int i = 0; i < 2; ++i) {
        cs.create_link_group(party_group[i]);
        cs.create_link_group(out_group[i]);
    }

    // --- witness allocation (struct field order; link groups per the
    //     #[link_groups] annotations) ---
    SettlementVars pv[2];
    for (int i = 0; i < 2; ++i) {
        auto alloc_list = [&](const std::vector<Fr>& vals, const char* group) {
            std::vector<Var> out;
            for (auto& v : vals) {
                Var x = cs.create_variable(v);
                if (group) cs.add_to_link_group(x, group);
                out.push_back(x);
            }
            return out;
        };
        auto obv = alloc_list(w.p[i].obligation.to_scalars(), nullptr);
        auto inv = alloc_list(w.p[i].intent.to_scalars(), party_group[i]);
        auto pam = alloc_list({w.p[i].pre_amount_share}, party_group[i]);
        auto ibv = alloc_list(w.p[i].input_balance.to_scalars(), party_group[i]);
        auto piv = alloc_list(w.p[i].pre_in_shares.to_scalars(), party_group[i]);
        auto obal = alloc_list(w.p[i].output_balance.to_scalars(), out_group[i]);
        auto pov = alloc_list(w.p[i].pre_out_shares.to_scalars(), out_group[i]);
        std::copy(obv.begin(), obv.end(), pv[i].ob.begin());
        std::copy(inv.begin(), inv.end(), pv[i].intent.begin());
        pv[i].pre_amount = pam[0];
        std::copy(ibv.begin(), ibv.end(), pv[i].in_bal.begin());
        std::copy(piv.begin(), piv.end(), pv[i].pre_in.begin());
        std::copy(obal.begin(), obal.end(), pv[i].out_bal.begin());
        std::copy(pov.begin(), pov.end(), pv[i].pre_out.begin());
    }
    // --- statement allocation (public inputs, field order) ---
    auto sscal = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : sscal) pub.push_back(cs.create_public_variable(s));
    Var new_amount[2] = {pub[0], pub[7]};
    std::array<Var, 3> new_in[2] = {{pub[1], pub[2], pub[3]}, {pub[8], pub[9], pub[10]}};
    std::array<Var, 3> new_out[2] = {{pub[4], pub[5], pub[6]}, {pub[11], pub[12], pub[13]}};
    Var relayer_fee[2] = {pub[14], pub[15]};
    Var protocol_fee = pub[16];

    // --- 1. obligation compatibility (:135-158) ---
    for (int i = 0; i < 2; ++i) {
        cs.enforce_in_range(pv[i].ob[2], AMOUNT_BITS);
        cs.enforce_in_range(pv[i].ob[3], AMOUNT_BITS);
    }
    cs.enforce_equal(pv[0].ob[0], pv[1].ob[1]);  // in0 == out1 token
    cs.enforce_equal(pv[0].ob[1], pv[1].ob[0]);
    cs.enforce_equal(pv[0].ob[2], pv[1].ob[3]);  // amount_in0 == amount_out1
    cs.enforce_equal(pv[0].ob[3], pv[1].ob[2]);

    for (int i = 0; i < 2; ++i) {
        // --- 2. fee take (:109-133) ---
        FeeTakeVars ft = fee_take_gadget(cs, pv[i].ob[3], relayer_fee[i], protocol_fee);
        // --- 3. intent constraints (settlement_lib.rs:46-79) ---
        cs.enforce_equal(pv[i].ob[0], pv[i].intent[0]);  // input_token == in_token
        cs.enforce_equal(pv[i].ob[1], pv[i].intent[1]);
        gte_gadget(cs, pv[i].intent[4], pv[i].ob[2], AMOUNT_BITS);
        Var min_out_fp = cs.mul(pv[i].intent[3], pv[i].ob[2]);  // mul_integer
        Var min_out = fp_floor_gadget(cs, min_out_fp);
        gte_gadget(cs, pv[i].ob[3], min_out, AMOUNT_BITS);
        // --- in-balance (:86-99) ---
        gte_gadget(cs, pv[i].in_bal[7], pv[i].ob[2], AMOUNT_BITS);
        // --- out-balance (:102-129) ---
        cs.enforce_equal(pv[i].out_bal[0], pv[i].ob[1]);     // mint == output token
        cs.enforce_equal(pv[i].out_bal[1], pv[i].intent[2]); // owner
        Var total_fee = cs.add(ft.relayer_fee, ft.protocol_fee);
        Var net_receive = cs.sub(pv[i].ob[3], total_fee);
        Var new_bal_amount = cs.add(pv[i].out_bal[7], net_receive);
        cs.enforce_in_range(new_bal_amount, AMOUNT_BITS);
        Var new_rfb = cs.add(pv[i].out_bal[5], ft.relayer_fee);
        Var new_pfb = cs.add(pv[i].out_bal[6], ft.protocol_fee);
        cs.enforce_in_range(new_rfb, AMOUNT_BITS);
        cs.enforce_in_range(new_pfb, AMOUNT_BITS);
        // --- 4. state updates (settlement_lib.rs:141-199) ---
        // share tuples are in PostMatchBalanceShare field order:
        // [0]=relayer_fee_balance, [1]=protocol_fee_balance, [2]=amount
        Var exp_amount = cs.sub(pv[i].pre_amount, pv[i].ob[2]);
        cs.enforce_equal(exp_amount, new_amount[i]);
        Var exp_in_amt = cs.sub(pv[i].pre_in[2], pv[i].ob[2]);
        cs.enforce_equal(exp_in_amt, new_in[i][2]);
        cs.enforce_equal(pv[i].pre_in[0], new_in[i][0]);
        cs.enforce_equal(pv[i].pre_in[1], new_in[i][1]);
        Var exp_out_amt = cs.add(pv[i].pre_out[2], net_receive);
        cs.enforce_equal(exp_out_amt, new_out[i][2]);
        Var exp_out_rfb = cs.add(pv[i].pre_out[0], ft.relayer_fee);
        cs.enforce_equal(exp_out_rfb, new_out[i][0]);
        Var exp_out_pfb = cs.add(pv[i].pre_out[1], ft.protocol_fee);
        cs.enforce_equal(exp_out_pfb, new_out[i][1]);
    }
}


// ========================= Valid Deposit =========================
// (zk_circuits/valid_deposit.rs — deposit into an EXISTING balance with a
//  full state rotation: Merkle opening of the old version, nullifier,
//  re-encrypted amount share, commitment to the new version.
//  MERKLE_HEIGHT = 10, constants/src/lib.rs:50.)

constexpr int MERKLE_HEIGHT = 10;

struct StateBalance {  // DarkpoolStateBalance native (20 scalars)
    Csprng recovery, share;
    Balance inner;
    Fr public_share[8];
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = {recovery.seed, Fr::from_u64(recovery.index), share.seed,
                             Fr::from_u64(share.index)};
        auto b = inner.to_scalars();
        v.insert(v.end(), b.begin(), b.end());
        v.insert(v.end(), public_share, public_share + 8);
        return v;
    }
};

struct VdWitness {  // ValidDepositWitness<10> (20 + 20 scalars)
    StateBalance old_balance;
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_indices[MERKLE_HEIGHT];
};
struct VdStatement {  // ValidDepositStatement (8 scalars)
    Deposit deposit;
    Fr merkle_root, old_nullifier, new_commitment, recovery_id, new_amount_share;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = deposit.to_scalars();
        v.push_back(merkle_root);
        v.push_back(old_nullifier);
        v.push_back(new_commitment);
        v.push_back(recovery_id);
        v.push_back(new_amount_share);
        return v;
    }
};

// fixed-seed builder mirroring test_helpers (valid_deposit.rs:199+)
inline void vd_build_witness_statement(uint64_t seed, VdWitness& w, VdStatement& st) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    Fr amount = Fr::from_u64(rng.next() & ((1ull << 50) - 1));
    w.old_balance.inner = {addr(), addr(), addr(), rng.fr(), rng.fr(),
                           Fr::from_u64(rng.next() & 0xFFFF),
                           Fr::from_u64(rng.next() & 0xFFFF),
                           Fr::from_u64(rng.next() & ((1ull << 50) - 1))};
    w.old_balance.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};  // index >= 1
    w.old_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int i = 0; i < 8; ++i) w.old_balance.public_share[i] = rng.fr();
    for (int i = 0; i < MERKLE_HEIGHT; ++i) {
        w.opening_elems[i] = rng.fr();
        w.opening_indices[i] = rng.next() & 1;
    }
    st.deposit = {w.old_balance.inner.owner, w.old_balance.inner.mint, amount};

    // old commitment + merkle root + nullifier
    auto priv_of = [](const Balance& inner, const Fr pub_[8]) {
        std::vector<Fr> p;
        auto iv = inner.to_scalars();
        for (int i = 0; i < 8; ++i) p.push_back(iv[i].sub(pub_[i]));
        return p;
    };
    std::vector<Fr> old_priv = priv_of(w.old_balance.inner, w.old_balance.public_share);
    std::vector<Fr> old_pub(w.old_balance.public_share, w.old_balance.public_share + 8);
    Fr old_comm = native_commitment(old_priv, w.old_balance.recovery, w.old_balance.share,
                                    old_pub);
    st.merkle_root = native_merkle_root(
        old_comm, std::vector<Fr>(w.opening_elems, w.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(w.opening_indices, w.opening_indices + MERKLE_HEIGHT));
    st.old_nullifier = native_nullifier(w.old_balance.recovery);

    // new balance: amount += deposit, re-encrypt amount, rotate streams
    StateBalance nb = w.old_balance;
    nb.inner.amount = nb.inner.amount.add(amount);
    Csprng share = nb.share;
    Fr pad = share.next();
    st.new_amount_share = nb.inner.amount.sub(pad);
    nb.public_share[7] = st.new_amount_share;
    nb.share = share;
    Csprng recovery = nb.recovery;
    st.recovery_id = recovery.next();
    nb.recovery = recovery;
    std::vector<Fr> new_priv = priv_of(nb.inner, nb.public_share);
    std::vector<Fr> new_pub(nb.public_share, nb.public_share + 8);
    st.new_commitment = native_commitment(new_priv, nb.recovery, nb.share, new_pub);
}

inline void vd_apply_constraints(PlonkCircuit& cs, const VdWitness& w,
                                 const VdStatement& st) {
    // --- witness allocation (field order) ---
    StateWrapperVars old_v;
    old_v.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                      cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    old_v.share = {cs.create_variable(w.old_balance.share.seed),
                   cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    auto iv = w.old_balance.inner.to_scalars();
    for (auto& s : iv) old_v.inner.push_back(cs.create_variable(s));
    for (int i = 0; i < 8; ++i)
        old_v.public_share.push_back(cs.create_variable(w.old_balance.public_share[i]));
    std::vector<Var> op_elems, op_idx;
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_elems.push_back(cs.create_variable(w.opening_elems[i]));
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_indices[i] ? Fr::one() : Fr::zero()));
    // --- statement (public) ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var dep_from = pub[0], dep_token = pub[1], dep_amount = pub[2];
    Var p_root = pub[3], p_null = pub[4], p_comm = pub[5], p_rid = pub[6],
        p_amt_share = pub[7];

    // validate_deposit (valid_deposit.rs:79-96)
    cs.enforce_in_range(dep_amount, 100);
    cs.enforce_equal(dep_token, old_v.inner[0]);
    cs.enforce_equal(dep_from, old_v.inner[1]);

    // complementary private shares (shares.rs:45-59)
    std::vector<Var> old_priv;
    for (int i = 0; i < 8; ++i)
        old_priv.push_back(cs.sub(old_v.inner[i], old_v.public_share[i]));

    // create_new_balance (:99-127)
    StateWrapperVars new_v = old_v;
    std::vector<Var> new_priv = old_priv;
    new_v.inner[7] = cs.add(old_v.inner[7], dep_amount);
    cs.enforce_in_range(new_v.inner[7], 100);
    std::vector<Var> pads, pubs_enc;
    {
        std::vector<Var> vals{new_v.inner[7]};
        stream_cipher_encrypt(cs, vals, new_v.share, pads, pubs_enc);
        new_priv[7] = pads[0];
        new_v.public_share[7] = pubs_enc[0];
        cs.enforce_equal(pubs_enc[0], p_amt_share);
    }

    // rotation (state_rotation.rs:83-122)
    Var rid = csprng_next(cs, new_v.recovery);  // RecoveryIdGadget
    cs.enforce_equal(rid, p_rid);
    Var old_comm = commitment_gadget(cs, old_priv, old_v.recovery, old_v.share,
                                     old_v.public_share);
    Var new_comm = commitment_gadget(cs, new_priv, new_v.recovery, new_v.share,
                                     new_v.public_share);
    cs.enforce_equal(new_comm, p_comm);
    Var root = merkle_root_gadget(cs, old_comm, op_elems, op_idx);
    cs.enforce_equal(root, p_root);
    Var nul = nullifier_gadget(cs, old_v);
    cs.enforce_equal(nul, p_null);
}



// ========================= Valid Withdrawal =========================
// (zk_circuits/valid_withdrawal.rs — withdraw from an existing balance:
//  nonzero amount <= balance, zero outstanding fees, full state rotation.)

struct VwStatement {  // ValidWithdrawalStatement (8 scalars)
    Fr to, token, amount;  // Withdrawal (withdrawal.rs:26-33)
    Fr merkle_root, old_nullifier, new_commitment, recovery_id, new_amount_share;
    std::vector<Fr> to_scalars() const {
        return {to, token, amount, merkle_root, old_nullifier, new_commitment,
                recovery_id, new_amount_share};
    }
};

inline void vw_build_witness_statement(uint64_t seed, VdWitness& w, VwStatement& st) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    uint64_t bal_amount = (rng.next() & ((1ull << 50) - 1)) + 2;
    uint64_t wd_amount = bal_amount / 2 + 1;  // nonzero, <= balance
    w.old_balance.inner = {addr(), addr(), addr(), rng.fr(), rng.fr(),
                           Fr::zero(), Fr::zero(),  // no outstanding fees
                           Fr::from_u64(bal_amount)};
    w.old_balance.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
    w.old_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int i = 0; i < 8; ++i) w.old_balance.public_share[i] = rng.fr();
    for (int i = 0; i < MERKLE_HEIGHT; ++i) {
        w.opening_elems[i] = rng.fr();
        w.opening_indices[i] = rng.next() & 1;
    }
    st.to = w.old_balance.inner.owner;
    st.token = w.old_balance.inner.mint;
    st.amount = Fr::from_u64(wd_amount);

    auto priv_of = [](const Balance& inner, const Fr pub_[8]) {
        std::vector<Fr> p;
        auto iv = inner.to_scalars();
        for (int i = 0; i < 8; ++i) p.push_back(iv[i].sub(pub_[i]));
        return p;
    };
    std::vector<Fr> old_priv = priv_of(w.old_balance.inner, w.old_balance.public_share);
    std::vector<Fr> old_pub(w.old_balance.public_share, w.old_balance.public_share + 8);
    Fr old_comm = native_commitment(old_priv, w.old_balance.recovery, w.old_balance.share,
                                    old_pub);
    st.merkle_root = native_merkle_root(
        old_comm, std::vector<Fr>(w.opening_elems, w.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(w.opening_indices, w.opening_indices + MERKLE_HEIGHT));
    st.old_nullifier = native_nullifier(w.old_balance.recovery);

    StateBalance nb = w.old_balance;
    nb.inner.amount = nb.inner.amount.sub(st.amount);
    Csprng share = nb.share;
    Fr pad = share.next();
    st.new_amount_share = nb.inner.amount.sub(pad);
    nb.public_share[7] = st.new_amount_share;
    nb.share = share;
    Csprng recovery = nb.recovery;
    st.recovery_id = recovery.next();
    nb.recovery = recovery;
    std::vector<Fr> new_priv = priv_of(nb.inner, nb.public_share);
    std::vector<Fr> new_pub(nb.public_share, nb.public_share + 8);
    st.new_commitment = native_commitment(new_priv, nb.recovery, nb.share, new_pub);
}

inline void vw_apply_constraints(PlonkCircuit& cs, const VdWitness& w,
                                 const VwStatement& st) {
    StateWrapperVars old_v;
    old_v.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                      cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    old_v.share = {cs.create_variable(w.old_balance.share.seed),
                   cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    auto iv = w.old_balance.inner.to_scalars();
    for (auto& s : iv) old_v.inner.push_back(cs.create_variable(s));
    for (int i = 0; i < 8; ++i)
        old_v.public_share.push_back(cs.create_variable(w.old_balance.public_share[i]));
    std::vector<Var> op_elems, op_idx;
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_elems.push_back(cs.create_variable(w.opening_elems[i]));
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_indices[i] ? Fr::one() : Fr::zero()));
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var wd_to = pub[0], wd_token = pub[1], wd_amount = pub[2];
    Var p_root = pub[3], p_null = pub[4], p_comm = pub[5], p_rid = pub[6],
        p_amt_share = pub[7];

    // validate_withdrawal (valid_withdrawal.rs:76-101)
    cs.enforce_in_range(wd_amount, 100);
    Var z = is_zero_gadget(cs, wd_amount);
    cs.enforce_false(z);
    cs.enforce_equal(wd_token, old_v.inner[0]);
    cs.enforce_equal(wd_to, old_v.inner[1]);
    gte_gadget(cs, old_v.inner[7], wd_amount, AMOUNT_BITS);
    // no outstanding fees (:103-114)
    cs.enforce_equal(old_v.inner[5], cs.zero());
    cs.enforce_equal(old_v.inner[6], cs.zero());

    std::vector<Var> old_priv;
    for (int i = 0; i < 8; ++i)
        old_priv.push_back(cs.sub(old_v.inner[i], old_v.public_share[i]));

    StateWrapperVars new_v = old_v;
    std::vector<Var> new_priv = old_priv;
    new_v.inner[7] = cs.sub(old_v.inner[7], wd_amount);
    std::vector<Var> pads, pubs_enc;
    {
        std::vector<Var> vals{new_v.inner[7]};
        stream_cipher_encrypt(cs, vals, new_v.share, pads, pubs_enc);
        new_priv[7] = pads[0];
        new_v.public_share[7] = pubs_enc[0];
        cs.enforce_equal(pubs_enc[0], p_amt_share);
    }
    Var rid = csprng_next(cs, new_v.recovery);
    cs.enforce_equal(rid, p_rid);
    Var old_comm = commitment_gadget(cs, old_priv, old_v.recovery, old_v.share,
                                     old_v.public_share);
    Var new_comm = commitment_gadget(cs, new_priv, new_v.recovery, new_v.share,
                                     new_v.public_share);
    cs.enforce_equal(new_comm, p_comm);
    Var root = merkle_root_gadget(cs, old_comm, op_elems, op_idx);
    cs.enforce_equal(root, p_root);
    Var nul = nullifier_gadget(cs, old_v);
    cs.enforce_equal(nul, p_null);
}


// ================== Intent And Balance Validity ==================
// (zk_circuits/validity_proofs/intent_and_balance.rs — the per-party
//  validity proof whose witness proof-links into the settlement proof's
//  intent_and_balance_settlement_party{0,1} groups.)

// intent_and_balance.rs:56 (IntentShare::NUM_SCALARS - 1: omit amount_in)
constexpr size_t INTENT_PARTIAL_COMMITMENT_SIZE = 4;
// intent_and_balance_first_fill.rs:53-54
// (DarkpoolBalance::NUM_SCALARS - PostMatchBalance::NUM_SCALARS = 8 - 3)
constexpr size_t BALANCE_PARTIAL_COMMITMENT_SIZE = 5;

struct StateIntent {  // DarkpoolStateIntent = StateWrapper<Intent> (14 scalars)
    Csprng recovery, share;
    Intent inner;
    Fr public_share[5];
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = {recovery.seed, Fr::from_u64(recovery.index), share.seed,
                             Fr::from_u64(share.index)};
        auto iv = inner.to_scalars();
        v.insert(v.end(), iv.begin(), iv.end());
        v.insert(v.end(), public_share, public_share + 5);
        return v;
    }
};

struct ValidityWitness {  // intent_and_balance.rs:236-272 (field order)
    StateIntent old_intent;
    Fr intent_opening_elems[MERKLE_HEIGHT];
    bool intent_opening_idx[MERKLE_HEIGHT];
    Intent intent;               // linked (denormalized from old_intent.inner)
    Fr new_amount_public_share;  // linked (re-encrypted amount_in share)
    StateBalance old_balance;
    Fr balance_opening_elems[MERKLE_HEIGHT];
    bool balance_opening_idx[MERKLE_HEIGHT];
    Balance balance;                      // linked
    PostMatchShare post_match_balance_shares;  // linked (rfb, pfb, amount order)
};

struct ValidityStatement {  // intent_and_balance.rs:277-309 (10 scalars)
    Fr intent_merkle_root, old_intent_nullifier;
    Fr intent_partial_private, intent_partial_public;  // PartialCommitment
    Fr intent_recovery_id;
    Fr balance_merkle_root, old_balance_nullifier;
    Fr balance_partial_private, balance_partial_public;
    Fr balance_recovery_id;
    std::vector<Fr> to_scalars() const {
        return {intent_merkle_root, old_intent_nullifier, intent_partial_private,
                intent_partial_public, intent_recovery_id, balance_merkle_root,
                old_balance_nullifier, balance_partial_private, balance_partial_public,
                balance_recovery_id};
    }
};

// native partial commitment (commitment.rs:149-181): commitment to the
// private shares + streams, and resumable commitment over the first K
// public shares.
inline void native_partial_commitment(const std::vector<Fr>& private_shares,
                                      const Csprng& recovery, const Csprng& share,
                                      const std::vector<Fr>& public_shares, size_t K,
                                      Fr& out_priv, Fr& out_pub) {
    std::vector<Fr> in = private_shares;
    in.push_back(recovery.seed);
    in.push_back(Fr::from_u64(recovery.index));
    in.push_back(share.seed);
    in.push_back(Fr::from_u64(share.index));
    out_priv = poseidon_hash(in.data(), in.size());
    Fr comm = public_shares[0];
    for (size_t i = 1; i < K; ++i) {
        Fr two[2] = {comm, public_shares[i]};
        comm = poseidon_hash(two, 2);
    }
    out_pub = comm;
}

// in-circuit partial commitment (commitment.rs:149-181)
inline std::pair<Var, Var> partial_commitment_gadget(
    PlonkCircuit& cs, const std::vector<Var>& private_share, const CsprngVar& recovery,
    const CsprngVar& share, const std::vector<Var>& public_share, size_t K) {
    PoseidonHashGadget h(cs);
    std::vector<Var> in = private_share;
    in.push_back(recovery.seed);
    in.push_back(recovery.index);
    in.push_back(share.seed);
    in.push_back(share.index);
    Var priv = h.hash(cs, in);
    std::vector<Var> firstK(public_share.begin(), public_share.begin() + K);
    Var pp = resumable_commitment(cs, firstK);
    return {priv, pp};
}

// ---- OUTPUT BALANCE VALIDITY (validity_proofs/output_balance.rs) ----
// The balance-only counterpart: rotate the party's OUTPUT balance with a
// partial commitment; links into the settlement's
// output_balance_settlement_party{0,1} groups (11 vars).

struct ObValidityWitness {  // output_balance.rs:137-155 (field order)
    StateBalance old_balance;
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_idx[MERKLE_HEIGHT];
    Balance balance;                           // linked (denormalized)
    PostMatchShare post_match_balance_shares;  // linked (rfb, pfb, amount)
};

struct ObValidityStatement {  // output_balance.rs:163-176 (5 scalars)
    Fr merkle_root, old_balance_nullifier;
    Fr partial_private, partial_public;  // PartialCommitment
    Fr recovery_id;
    std::vector<Fr> to_scalars() const {
        return {merkle_root, old_balance_nullifier, partial_private, partial_public,
                recovery_id};
    }
};

// ================== Intent And Balance Public Settlement ==================
// (settlement/intent_and_balance_public_settlement.rs — single-party public
//  settlement by a relayer cluster: the obligation and pre-update shares are
//  LEAKED in the statement and applied on-chain; links against the party-0
//  groups of the validity proofs at the private settlement's layout.)

struct PubSettlementStatement {  // :148-180 (14 scalars)
    Obligation obligation;
    Fr amount_public_share;
    PostMatchShare in_shares, out_shares;
    Fr relayer_fee_repr, protocol_fee_repr;  // FeeRates (fee.rs:43-48)
    Fr relayer_fee_recipient;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = obligation.to_scalars();
        v.push_back(amount_public_share);
        auto a = in_shares.to_scalars();
        v.insert(v.end(), a.begin(), a.end());
        auto b = out_shares.to_scalars();
        v.insert(v.end(), b.begin(), b.end());
        v.push_back(relayer_fee_repr);
        v.push_back(protocol_fee_repr);
        v.push_back(relayer_fee_recipient);
        return v;
    }
};

// witness = the party's linked vars (SettlementParty minus the obligation,
// :100-143); built from bundle party 0 so its links match the validity
// proofs.  Groups are placed at the private settlement's layout
// (proof_linking_groups, :194-213).
inline void pub_settlement_apply_constraints(PlonkCircuit& cs,
                                             const SettlementParty& w,
                                             const PubSettlementStatement& st,
                                             int alignment, int64_t pg_offset,
                                             int64_t og_offset) {
    const char* pg = "intent_and_balance_settlement_party0";
    const char* og = "output_balance_settlement_party0";
    cs.create_link_group(pg, alignment, pg_offset);
    cs.create_link_group(og, alignment, og_offset);
    auto alloc_list = [&](const std::vector<Fr>& vals, const char* group) {
        std::vector<Var> out;
        for (auto& v : vals) {
            Var x = cs.create_variable(v);
            if (group) cs.add_to_link_group(x, group);
            out.push_back(x);
        }
        return out;
    };
    // --- witness (field order :100-143; linked groups per annotations) ---
    auto intent_v = alloc_list(w.intent.to_scalars(), pg);
    auto pre_amt = alloc_list({w.pre_amount_share}, pg);
    auto in_bal = alloc_list(w.input_balance.to_scalars(), pg);
    auto pre_in = alloc_list(w.pre_in_shares.to_scalars(), pg);
    auto out_bal = alloc_list(w.output_balance.to_scalars(), og);
    auto pre_out = alloc_list(w.pre_out_shares.to_scalars(), og);

    // --- statement (public inputs, field order) ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    std::array<Var, 4> ob{pub[0], pub[1], pub[2], pub[3]};
    Var p_amt_share = pub[4];
    std::array<Var, 3> p_in{pub[5], pub[6], pub[7]};
    std::array<Var, 3> p_out{pub[8], pub[9], pub[10]};
    Var relayer_rate = pub[11], protocol_rate = pub[12], p_fee_recipient = pub[13];

    // 1. fee take from the statement rates (:52-56)
    FeeTakeVars ft = fee_take_gadget(cs, ob[3], relayer_rate, protocol_rate);
    // 2. intent/balance obligation constraints (settlement_lib.rs:29-126)
    cs.enforce_equal(ob[0], intent_v[0]);
    cs.enforce_equal(ob[1], intent_v[1]);
    gte_gadget(cs, intent_v[4], ob[2], AMOUNT_BITS);
    Var min_out_fp = cs.mul(intent_v[3], ob[2]);
    Var min_out = fp_floor_gadget(cs, min_out_fp);
    gte_gadget(cs, ob[3], min_out, AMOUNT_BITS);
    gte_gadget(cs, in_bal[7], ob[2], AMOUNT_BITS);
    cs.enforce_equal(out_bal[0], ob[1]);
    cs.enforce_equal(out_bal[1], intent_v[2]);
    Var total_fee = cs.add(ft.relayer_fee, ft.protocol_fee);
    Var net_receive = cs.sub(ob[3], total_fee);
    Var new_bal_amount = cs.add(out_bal[7], net_receive);
    cs.enforce_in_range(new_bal_amount, AMOUNT_BITS);
    Var new_rfb = cs.add(out_bal[5], ft.relayer_fee);
    Var new_pfb = cs.add(out_bal[6], ft.protocol_fee);
    cs.enforce_in_range(new_rfb, AMOUNT_BITS);
    cs.enforce_in_range(new_pfb, AMOUNT_BITS);
    // 3. leaked pre-update shares match the proof-linked witness (:68-86)
    cs.enforce_equal(pre_amt[0], p_amt_share);
    for (int k = 0; k < 3; ++k) cs.enforce_equal(pre_in[k], p_in[k]);
    for (int k = 0; k < 3; ++k) cs.enforce_equal(pre_out[k], p_out[k]);
    // 4. relayer fee recipient leak (:88-93)
    cs.enforce_equal(out_bal[2], p_fee_recipient);
}

// ================== Intent Only Settlement + Validity ==================
// (settlement/intent_only_public_settlement.rs + validity_proofs/intent_only.rs
//  — a private intent capitalized by a PUBLIC balance: the settlement leaks
//  the obligation; one 5-var link group "intent_only_settlement" placed by
//  the settlement circuit and inherited by the validity circuit.)

struct IoSettlementStatement {  // intent_only_public_settlement.rs:85-112 (6)
    Obligation obligation;
    Fr relayer_fee_repr;      // transcript-only (non-malleable), unconstrained
    Fr relayer_fee_recipient;  // likewise
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = obligation.to_scalars();
        v.push_back(relayer_fee_repr);
        v.push_back(relayer_fee_recipient);
        return v;
    }
};

struct IoValidityWitness {  // intent_only.rs:148-166
    StateIntent old_intent;
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_idx[MERKLE_HEIGHT];
    Intent intent;  // linked (denormalized new-intent inner)
};
struct IoValidityStatement {  // intent_only.rs:174-204 (7 scalars)
    Fr owner, merkle_root, old_intent_nullifier, new_amount_public_share;
    Fr partial_private, partial_public;
    Fr recovery_id;
    std::vector<Fr> to_scalars() const {
        return {owner, merkle_root, old_intent_nullifier, new_amount_public_share,
                partial_private, partial_public, recovery_id};
    }
};

// consistent pair: the validity witness wraps the same intent the settlement
// witness links, and the obligation respects the intent's constraints
inline void io_bundle_build(uint64_t seed, IoValidityWitness& vw, IoValidityStatement& vs,
                            IoSettlementStatement& ss) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    Fr token0 = addr(), token1 = addr(), owner = addr();
    uint64_t amount = (rng.next() & ((1ull << 52) - 1)) + 2;
    uint64_t traded_in = amount / 2 + 1;
    uint64_t traded_out = (rng.next() & ((1ull << 52) - 1)) + 2;
    auto half_price_repr = [&](uint64_t out_amt, uint64_t in_amt) {
        unsigned __int128 r = ((unsigned __int128)out_amt << 63) / in_amt / 2;
        u64 l[4] = {(u64)r, (u64)(r >> 64), 0, 0};
        return Fr::from_canonical(l);
    };
    vw.intent = {token0, token1, owner, half_price_repr(traded_out, traded_in),
                 Fr::from_u64(amount)};
    ss.obligation = {token0, token1, Fr::from_u64(traded_in), Fr::from_u64(traded_out)};
    auto fee_repr = [&]() {
        unsigned __int128 r = ((unsigned __int128)1 << 63) / (1000 + (rng.next() & 1023));
        u64 l[4] = {(u64)r, (u64)(r >> 64), 0, 0};
        return Fr::from_canonical(l);
    };
    ss.relayer_fee_repr = fee_repr();
    ss.relayer_fee_recipient = addr();

    vw.old_intent.inner = vw.intent;
    vw.old_intent.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
    vw.old_intent.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int k = 0; k < 5; ++k) vw.old_intent.public_share[k] = rng.fr();
    for (int k = 0; k < MERKLE_HEIGHT; ++k) {
        vw.opening_elems[k] = rng.fr();
        vw.opening_idx[k] = rng.next() & 1;
    }
    auto iv = vw.old_intent.inner.to_scalars();
    std::vector<Fr> old_priv;
    for (int k = 0; k < 5; ++k)
        old_priv.push_back(iv[k].sub(vw.old_intent.public_share[k]));
    Fr old_comm = native_commitment(
        old_priv, vw.old_intent.recovery, vw.old_intent.share,
        std::vector<Fr>(vw.old_intent.public_share, vw.old_intent.public_share + 5));
    vs.owner = owner;
    vs.merkle_root = native_merkle_root(
        old_comm, std::vector<Fr>(vw.opening_elems, vw.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(vw.opening_idx, vw.opening_idx + MERKLE_HEIGHT));
    vs.old_intent_nullifier = native_nullifier(vw.old_intent.recovery);
    StateIntent ni = vw.old_intent;
    Fr pad = ni.share.next();
    vs.new_amount_public_share = ni.inner.amount_in.sub(pad);
    ni.public_share[4] = vs.new_amount_public_share;
    std::vector<Fr> new_priv = old_priv;
    new_priv[4] = pad;
    vs.recovery_id = ni.recovery.next();
    native_partial_commitment(
        new_priv, ni.recovery, ni.share,
        std::vector<Fr>(ni.public_share, ni.public_share + 5),
        INTENT_PARTIAL_COMMITMENT_SIZE, vs.partial_private, vs.partial_public);
}

// intent_only_public_settlement.rs:48-62: intent constraints only; the group
// is PLACED by this circuit (auto layout)
inline void io_settlement_apply_constraints(PlonkCircuit& cs, const Intent& w,
                                            const IoSettlementStatement& st) {
    cs.create_link_group("intent_only_settlement", -1, -1);
    std::array<Var, 5> intent_v;
    auto iv = w.to_scalars();
    for (int k = 0; k < 5; ++k) {
        intent_v[k] = cs.create_variable(iv[k]);
        cs.add_to_link_group(intent_v[k], "intent_only_settlement");
    }
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    // verify_intent_constraints (settlement_lib.rs:45-74)
    cs.enforce_equal(pub[0], intent_v[0]);
    cs.enforce_equal(pub[1], intent_v[1]);
    gte_gadget(cs, intent_v[4], pub[2], AMOUNT_BITS);
    Var min_out_fp = cs.mul(intent_v[3], pub[2]);
    Var min_out = fp_floor_gadget(cs, min_out_fp);
    gte_gadget(cs, pub[3], min_out, AMOUNT_BITS);
}

// intent_only.rs:63-116; the link group inherits the settlement's placement
inline void io_validity_apply_constraints(PlonkCircuit& cs, const IoValidityWitness& w,
                                          const IoValidityStatement& st, int alignment,
                                          int64_t offset) {
    const char* g = "intent_only_settlement";
    cs.create_link_group(g, alignment, offset);

    // --- witness (field order) ---
    StateWrapperVars oi;
    oi.recovery = {cs.create_variable(w.old_intent.recovery.seed),
                   cs.create_variable(Fr::from_u64(w.old_intent.recovery.index))};
    oi.share = {cs.create_variable(w.old_intent.share.seed),
                cs.create_variable(Fr::from_u64(w.old_intent.share.index))};
    for (auto& s : w.old_intent.inner.to_scalars())
        oi.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 5; ++k)
        oi.public_share.push_back(cs.create_variable(w.old_intent.public_share[k]));
    std::vector<Var> op_elems, op_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_elems.push_back(cs.create_variable(w.opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_idx[k] ? Fr::one() : Fr::zero()));
    std::array<Var, 5> intent_v;
    {
        auto iv = w.intent.to_scalars();
        for (int k = 0; k < 5; ++k) {
            intent_v[k] = cs.create_variable(iv[k]);
            cs.add_to_link_group(intent_v[k], g);
        }
    }
    // --- statement ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var p_owner = pub[0], p_root = pub[1], p_null = pub[2], p_amt = pub[3],
        p_priv = pub[4], p_pub = pub[5], p_rid = pub[6];

    // 1-2. complementary shares + new intent (:69-79, :119-143)
    std::vector<Var> old_priv;
    for (int k = 0; k < 5; ++k)
        old_priv.push_back(cs.sub(oi.inner[k], oi.public_share[k]));
    CsprngVar ni_share = oi.share;
    std::vector<Var> pads, cipher;
    stream_cipher_encrypt(cs, {oi.inner[4]}, ni_share, pads, cipher);
    std::vector<Var> new_priv = old_priv;
    new_priv[4] = pads[0];
    std::vector<Var> new_pub = oi.public_share;
    new_pub[4] = cipher[0];
    // 3. denormalized intent + leaks (:81-92)
    for (int k = 0; k < 5; ++k) cs.enforce_equal(oi.inner[k], intent_v[k]);
    cs.enforce_equal(cipher[0], p_amt);
    cs.enforce_equal(oi.inner[2], p_owner);
    // 4. rotation with partial commitment (:94-113)
    CsprngVar ni_rec = oi.recovery;
    Var rid = csprng_next(cs, ni_rec);
    cs.enforce_equal(rid, p_rid);
    auto pc = partial_commitment_gadget(cs, new_priv, ni_rec, ni_share, new_pub,
                                        INTENT_PARTIAL_COMMITMENT_SIZE);
    cs.enforce_equal(pc.first, p_priv);
    cs.enforce_equal(pc.second, p_pub);
    Var old_comm = commitment_gadget(cs, old_priv, oi.recovery, oi.share,
                                     oi.public_share);
    Var root = merkle_root_gadget(cs, old_comm, op_elems, op_idx);
    cs.enforce_equal(root, p_root);
    Var nul = nullifier_gadget(cs, oi);
    cs.enforce_equal(nul, p_null);
}

// ================== Bounded Settlements ==================
// (settlement/intent_only_bounded_settlement.rs +
//  intent_and_balance_bounded_settlement.rs — the trade size is chosen at
//  runtime by an external party within [min, max]; constraints bound the
//  WORST CASE: max amount in, price lower bound, max output overflow.)

constexpr int PRICE_BITS = FP_PRECISION + 64;  // circuit-types/src/lib.rs:64

struct BoundedMatchResult {  // bounded_match_result.rs:41-58 (6 scalars)
    Fr internal_party_input_token, internal_party_output_token;
    Fr min_internal_party_amount_in, max_internal_party_amount_in;
    Fr price_repr;
    Fr block_deadline;
    std::vector<Fr> to_scalars() const {
        return {internal_party_input_token, internal_party_output_token,
                min_internal_party_amount_in, max_internal_party_amount_in,
                price_repr, block_deadline};
    }
};

struct IoBoundedStatement {  // intent_only_bounded_settlement.rs:88-114 (9)
    BoundedMatchResult bmr;
    Fr internal_relayer_fee_repr, external_relayer_fee_repr, relayer_fee_recipient;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = bmr.to_scalars();
        v.push_back(internal_relayer_fee_repr);
        v.push_back(external_relayer_fee_repr);
        v.push_back(relayer_fee_recipient);
        return v;
    }
};

struct IbBoundedStatement {  // intent_and_balance_bounded_settlement.rs:141-182
    BoundedMatchResult bmr;   // (16 scalars)
    Fr amount_public_share;
    PostMatchShare in_shares, out_shares;
    Fr internal_relayer_fee_repr, external_relayer_fee_repr, relayer_fee_recipient;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = bmr.to_scalars();
        v.push_back(amount_public_share);
        auto a = in_shares.to_scalars();
        v.insert(v.end(), a.begin(), a.end());
        auto b = out_shares.to_scalars();
        v.insert(v.end(), b.begin(), b.end());
        v.push_back(internal_relayer_fee_repr);
        v.push_back(external_relayer_fee_repr);
        v.push_back(relayer_fee_recipient);
        return v;
    }
};

// BoundedSettlementGadget::verify_intent_constraints (settlement_lib.rs:224-261)
inline void bounded_intent_constraints(PlonkCircuit& cs,
                                       const std::array<Var, 5>& intent_v,
                                       const std::array<Var, 6>& bmr) {
    cs.enforce_equal(bmr[0], intent_v[0]);
    cs.enforce_equal(bmr[1], intent_v[1]);
    gte_gadget(cs, intent_v[4], bmr[3], AMOUNT_BITS);  // amount_in >= max_in
    gte_gadget(cs, bmr[4], intent_v[3], PRICE_BITS);   // price >= min_price
}

// intent_only_bounded_settlement.rs:52-67 (group inherits the intent-only
// public settlement placement)
inline void io_bounded_apply_constraints(PlonkCircuit& cs, const Intent& w,
                                         const IoBoundedStatement& st, int alignment,
                                         int64_t offset) {
    cs.create_link_group("intent_only_settlement", alignment, offset);
    std::array<Var, 5> intent_v;
    auto iv = w.to_scalars();
    for (int k = 0; k < 5; ++k) {
        intent_v[k] = cs.create_variable(iv[k]);
        cs.add_to_link_group(intent_v[k], "intent_only_settlement");
    }
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    std::array<Var, 6> bmr{pub[0], pub[1], pub[2], pub[3], pub[4], pub[5]};
    bounded_intent_constraints(cs, intent_v, bmr);
}

// intent_and_balance_bounded_settlement.rs:45-86 (groups inherit the private
// settlement party-0 placement)
inline void ib_bounded_apply_constraints(PlonkCircuit& cs, const SettlementParty& w,
                                         const IbBoundedStatement& st, int alignment,
                                         int64_t pg_offset, int64_t og_offset) {
    const char* pg = "intent_and_balance_settlement_party0";
    const char* og = "output_balance_settlement_party0";
    cs.create_link_group(pg, alignment, pg_offset);
    cs.create_link_group(og, alignment, og_offset);
    auto alloc_list = [&](const std::vector<Fr>& vals, const char* group) {
        std::vector<Var> out;
        for (auto& v : vals) {
            Var x = cs.create_variable(v);
            if (group) cs.add_to_link_group(x, group);
            out.push_back(x);
        }
        return out;
    };
    auto intent_l = alloc_list(w.intent.to_scalars(), pg);
    auto pre_amt = alloc_list({w.pre_amount_share}, pg);
    auto in_bal = alloc_list(w.input_balance.to_scalars(), pg);
    auto pre_in = alloc_list(w.pre_in_shares.to_scalars(), pg);
    auto out_bal = alloc_list(w.output_balance.to_scalars(), og);
    auto pre_out = alloc_list(w.pre_out_shares.to_scalars(), og);
    std::array<Var, 5> intent_v;
    std::copy(intent_l.begin(), intent_l.end(), intent_v.begin());

    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    std::array<Var, 6> bmr{pub[0], pub[1], pub[2], pub[3], pub[4], pub[5]};
    Var p_amt_share = pub[6];
    std::array<Var, 3> p_in{pub[7], pub[8], pub[9]};
    std::array<Var, 3> p_out{pub[10], pub[11], pub[12]};
    Var p_fee_recipient = pub[15];

    // 1. bounded match-result constraints (settlement_lib.rs:208-320)
    bounded_intent_constraints(cs, intent_v, bmr);
    gte_gadget(cs, in_bal[7], bmr[3], AMOUNT_BITS);  // balance >= max_in
    cs.enforce_equal(out_bal[0], bmr[1]);            // mint == output token
    cs.enforce_equal(out_bal[1], intent_v[2]);       // owner
    Var max_out_fp = cs.mul(bmr[4], bmr[3]);         // price * max_in
    Var max_out = fp_floor_gadget(cs, max_out_fp);
    Var max_bal_amount = cs.add(out_bal[7], max_out);
    cs.enforce_in_range(max_bal_amount, AMOUNT_BITS);
    // 2. leaked pre-update shares (:61-78)
    cs.enforce_equal(pre_amt[0], p_amt_share);
    for (int k = 0; k < 3; ++k) cs.enforce_equal(pre_in[k], p_in[k]);
    for (int k = 0; k < 3; ++k) cs.enforce_equal(pre_out[k], p_out[k]);
    // 3. relayer fee recipient leak (:80-85)
    cs.enforce_equal(out_bal[2], p_fee_recipient);
}

// consistent bounded statements from the bundles
inline void io_bounded_statement_build(uint64_t seed, const IoValidityWitness& vw,
                                       const IoSettlementStatement& ss,
                                       IoBoundedStatement& st) {
    Lcg rng(seed ^ 0xB0DEDB0DEDB0DEDull);
    u64 max_l[4], min_l[4];
    ss.obligation.amount_in.to_canonical(max_l);
    st.bmr.internal_party_input_token = vw.intent.in_token;
    st.bmr.internal_party_output_token = vw.intent.out_token;
    st.bmr.max_internal_party_amount_in = ss.obligation.amount_in;
    min_l[0] = max_l[0] / 2 + 1;
    st.bmr.min_internal_party_amount_in = Fr::from_u64(min_l[0]);
    st.bmr.price_repr = vw.intent.min_price_repr.dbl();  // >= min_price
    st.bmr.block_deadline = Fr::from_u64(rng.next() & 0xFFFFFFFF);
    st.internal_relayer_fee_repr = ss.relayer_fee_repr;
    st.external_relayer_fee_repr = ss.relayer_fee_repr;
    st.relayer_fee_recipient = ss.relayer_fee_recipient;
}

// ---- INTENT ONLY FIRST FILL VALIDITY (intent_only_first_fill.rs) ----
// First fill: the intent is NOT yet in the tree; prove well-formedness and
// commit to the private shares (the contract absorbs the public shares).

struct IoffWitness {  // :105-123 (field order)
    Intent intent;  // linked "intent_only_settlement"
    Csprng share_stream, recovery_stream;
    Fr private_shares[5];
};
struct IoffStatement {  // :132-149 (8 scalars)
    Fr owner, intent_private_commitment, recovery_id;
    Fr intent_public_share[5];
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = {owner, intent_private_commitment, recovery_id};
        v.insert(v.end(), intent_public_share, intent_public_share + 5);
        return v;
    }
};

// uses the SAME intent as io_bundle_build(seed), so the first-fill proof
// links against rng_circ_build_io_settlement(seed)'s proof
inline void ioff_build(uint64_t seed, IoffWitness& w, IoffStatement& st) {
    {
        IoValidityWitness vw;
        IoValidityStatement vs;
        IoSettlementStatement ss;
        io_bundle_build(seed, vw, vs, ss);
        w.intent = vw.intent;
    }
    Lcg rng(seed ^ 0xF157F111F157F111ull);
    w.share_stream = {rng.fr(), 0};
    w.recovery_stream = {rng.fr(), 0};
    for (int k = 0; k < 5; ++k) w.private_shares[k] = rng.fr();
    st.owner = w.intent.owner;
    auto iv = w.intent.to_scalars();
    for (int k = 0; k < 5; ++k)
        st.intent_public_share[k] = iv[k].sub(w.private_shares[k]);
    Csprng rec = w.recovery_stream;
    st.recovery_id = rec.next();
    std::vector<Fr> in(w.private_shares, w.private_shares + 5);
    in.push_back(rec.seed);
    in.push_back(Fr::from_u64(rec.index));
    in.push_back(w.share_stream.seed);
    in.push_back(Fr::from_u64(w.share_stream.index));
    st.intent_private_commitment = poseidon_hash(in.data(), in.size());
}

inline void ioff_apply_constraints(PlonkCircuit& cs, const IoffWitness& w,
                                   const IoffStatement& st, int alignment,
                                   int64_t offset) {
    const char* g = "intent_only_settlement";
    cs.create_link_group(g, alignment, offset);
    // --- witness (field order) ---
    std::array<Var, 5> intent_v;
    auto iv = w.intent.to_scalars();
    for (int k = 0; k < 5; ++k) {
        intent_v[k] = cs.create_variable(iv[k]);
        cs.add_to_link_group(intent_v[k], g);
    }
    CsprngVar share{cs.create_variable(w.share_stream.seed),
                    cs.create_variable(Fr::from_u64(w.share_stream.index))};
    CsprngVar rec{cs.create_variable(w.recovery_stream.seed),
                  cs.create_variable(Fr::from_u64(w.recovery_stream.index))};
    std::vector<Var> priv;
    for (int k = 0; k < 5; ++k) priv.push_back(cs.create_variable(w.private_shares[k]));
    // --- statement ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));

    // build_and_validate_intent (:68-96)
    cs.enforce_in_range(intent_v[4], AMOUNT_BITS);
    cs.enforce_in_range(intent_v[3], PRICE_BITS);
    cs.enforce_equal(intent_v[2], pub[0]);
    for (int k = 0; k < 5; ++k) {
        Var share_k = cs.sub(intent_v[k], priv[k]);
        cs.enforce_equal(share_k, pub[3 + k]);
    }
    // recovery id over the wrapper's recovery stream (:56-58)
    Var rid = csprng_next(cs, rec);
    cs.enforce_equal(rid, pub[2]);
    // private commitment with the ADVANCED recovery stream (:60-62)
    PoseidonHashGadget h(cs);
    std::vector<Var> in = priv;
    in.push_back(rec.seed);
    in.push_back(rec.index);
    in.push_back(share.seed);
    in.push_back(share.index);
    Var pc = h.hash(cs, in);
    cs.enforce_equal(pc, pub[1]);
}

// ================== Valid Order Cancellation ==================
// (zk_circuits/valid_order_cancellation.rs — prove the intent exists and
//  spend its nullifier; the owner is leaked for contract authorization.)

struct VocWitness {  // valid_order_cancellation.rs:77-84
    StateIntent old_intent;
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_idx[MERKLE_HEIGHT];
};
struct VocStatement {  // :95-105 (3 scalars)
    Fr merkle_root, old_intent_nullifier, owner;
    std::vector<Fr> to_scalars() const {
        return {merkle_root, old_intent_nullifier, owner};
    }
};

inline void voc_build_witness_statement(uint64_t seed, VocWitness& w, VocStatement& st) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    Fr owner = addr();
    w.old_intent.inner = {addr(), addr(), owner, rng.fr(), Fr::from_u64(rng.next())};
    w.old_intent.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
    w.old_intent.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int k = 0; k < 5; ++k) w.old_intent.public_share[k] = rng.fr();
    for (int k = 0; k < MERKLE_HEIGHT; ++k) {
        w.opening_elems[k] = rng.fr();
        w.opening_idx[k] = rng.next() & 1;
    }
    auto iv = w.old_intent.inner.to_scalars();
    std::vector<Fr> priv;
    for (int k = 0; k < 5; ++k) priv.push_back(iv[k].sub(w.old_intent.public_share[k]));
    Fr comm = native_commitment(
        priv, w.old_intent.recovery, w.old_intent.share,
        std::vector<Fr>(w.old_intent.public_share, w.old_intent.public_share + 5));
    st.merkle_root = native_merkle_root(
        comm, std::vector<Fr>(w.opening_elems, w.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(w.opening_idx, w.opening_idx + MERKLE_HEIGHT));
    st.old_intent_nullifier = native_nullifier(w.old_intent.recovery);
    st.owner = owner;
}

inline void voc_apply_constraints(PlonkCircuit& cs, const VocWitness& w,
                                  const VocStatement& st) {
    StateWrapperVars oi;
    oi.recovery = {cs.create_variable(w.old_intent.recovery.seed),
                   cs.create_variable(Fr::from_u64(w.old_intent.recovery.index))};
    oi.share = {cs.create_variable(w.old_intent.share.seed),
                cs.create_variable(Fr::from_u64(w.old_intent.share.index))};
    for (auto& s : w.old_intent.inner.to_scalars())
        oi.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 5; ++k)
        oi.public_share.push_back(cs.create_variable(w.old_intent.public_share[k]));
    std::vector<Var> op_elems, op_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_elems.push_back(cs.create_variable(w.opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_idx[k] ? Fr::one() : Fr::zero()));
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));

    // 1. intent exists in the tree (:48-60)
    std::vector<Var> priv;
    for (int k = 0; k < 5; ++k)
        priv.push_back(cs.sub(oi.inner[k], oi.public_share[k]));
    Var comm = commitment_gadget(cs, priv, oi.recovery, oi.share, oi.public_share);
    Var root = merkle_root_gadget(cs, comm, op_elems, op_idx);
    cs.enforce_equal(root, pub[0]);
    // 2. nullifier (:62-64)
    Var nul = nullifier_gadget(cs, oi);
    cs.enforce_equal(nul, pub[1]);
    // 3. owner leak (:66-67)
    cs.enforce_equal(oi.inner[2], pub[2]);
}

// The full bundle: one settlement witness/statement plus, per party, the
// INTENT AND BALANCE VALIDITY and OUTPUT BALANCE VALIDITY
// witnesses/statements — mutually consistent so all four validity proofs
// link into the settlement proof (the production proof bundle the reference
// relayer submits: 1 settlement proof + 4 link proofs,
// native_proof_manager.rs:554-590).
struct ValidityBundle {
    SettlementWitness sw;
    SettlementStatement sst;
    ValidityWitness vw[2];
    ValidityStatement vst[2];
    ObValidityWitness ow[2];
    ObValidityStatement ost[2];
    // Schnorr secret keys whose public keys are the input balances'
    // `authority` fields (needed by the first-fill circuits that bootstrap
    // intent authorization from the balance)
    JjScalar auth_sk[2];
};

inline JjScalar jj_random_scalar(Lcg& rng) {
    JjScalar s{{rng.next() | (rng.next() << 52), rng.next() | (rng.next() << 52),
                rng.next() | (rng.next() << 52), rng.next() & 0x3FFFFFFFFFFull}};
    return s;  // < 2^250 < l
}

inline void validity_bundle_build(uint64_t seed, ValidityBundle& b) {
    settlement_build_witness_statement(seed, b.sw, b.sst);
    Lcg rng(seed ^ 0x9E3779B97F4A7C15ull);
    // give the input balances REAL Schnorr authorities (first-fill circuits
    // verify signatures under these keys)
    for (int i = 0; i < 2; ++i) {
        b.auth_sk[i] = jj_random_scalar(rng);
        JjPoint vk = jj_pubkey(b.auth_sk[i]);
        b.sw.p[i].input_balance.authority_x = vk.x;
        b.sw.p[i].input_balance.authority_y = vk.y;
    }
    auto priv_of = [](const std::vector<Fr>& inner, const Fr* pub_, size_t n) {
        std::vector<Fr> p;
        for (size_t i = 0; i < n; ++i) p.push_back(inner[i].sub(pub_[i]));
        return p;
    };
    for (int i = 0; i < 2; ++i) {
        ValidityWitness& v = b.vw[i];
        ValidityStatement& st = b.vst[i];
        // old intent state element wrapping the party's intent
        v.old_intent.inner = b.sw.p[i].intent;
        v.old_intent.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
        v.old_intent.share = {rng.fr(), rng.next() & 0xFFFFFF};
        for (int k = 0; k < 5; ++k) v.old_intent.public_share[k] = rng.fr();
        for (int k = 0; k < MERKLE_HEIGHT; ++k) {
            v.intent_opening_elems[k] = rng.fr();
            v.intent_opening_idx[k] = rng.next() & 1;
        }
        v.intent = b.sw.p[i].intent;
        // old balance state element wrapping the party's input balance
        v.old_balance.inner = b.sw.p[i].input_balance;
        v.old_balance.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
        v.old_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
        for (int k = 0; k < 8; ++k) v.old_balance.public_share[k] = rng.fr();
        for (int k = 0; k < MERKLE_HEIGHT; ++k) {
            v.balance_opening_elems[k] = rng.fr();
            v.balance_opening_idx[k] = rng.next() & 1;
        }
        v.balance = b.sw.p[i].input_balance;

        // --- intent rotation natives (build_new_intent :135-158) ---
        auto old_iv = v.old_intent.inner.to_scalars();
        std::vector<Fr> old_priv_i = priv_of(old_iv, v.old_intent.public_share, 5);
        Fr old_comm_i = native_commitment(
            old_priv_i, v.old_intent.recovery, v.old_intent.share,
            std::vector<Fr>(v.old_intent.public_share, v.old_intent.public_share + 5));
        st.intent_merkle_root = native_merkle_root(
            old_comm_i,
            std::vector<Fr>(v.intent_opening_elems, v.intent_opening_elems + MERKLE_HEIGHT),
            std::vector<bool>(v.intent_opening_idx, v.intent_opening_idx + MERKLE_HEIGHT));
        st.old_intent_nullifier = native_nullifier(v.old_intent.recovery);
        StateIntent ni = v.old_intent;
        Fr pad = ni.share.next();
        v.new_amount_public_share = ni.inner.amount_in.sub(pad);
        ni.public_share[4] = v.new_amount_public_share;
        std::vector<Fr> new_priv_i = old_priv_i;
        new_priv_i[4] = pad;
        st.intent_recovery_id = ni.recovery.next();
        native_partial_commitment(
            new_priv_i, ni.recovery, ni.share,
            std::vector<Fr>(ni.public_share, ni.public_share + 5),
            INTENT_PARTIAL_COMMITMENT_SIZE, st.intent_partial_private,
            st.intent_partial_public);

        // --- balance rotation natives (build_new_balance :199-238) ---
        auto old_bv = v.old_balance.inner.to_scalars();
        std::vector<Fr> old_priv_b = priv_of(old_bv, v.old_balance.public_share, 8);
        Fr old_comm_b = native_commitment(
            old_priv_b, v.old_balance.recovery, v.old_balance.share,
            std::vector<Fr>(v.old_balance.public_share, v.old_balance.public_share + 8));
        st.balance_merkle_root = native_merkle_root(
            old_comm_b,
            std::vector<Fr>(v.balance_opening_elems, v.balance_opening_elems + MERKLE_HEIGHT),
            std::vector<bool>(v.balance_opening_idx, v.balance_opening_idx + MERKLE_HEIGHT));
        st.old_balance_nullifier = native_nullifier(v.old_balance.recovery);
        StateBalance nb = v.old_balance;
        // re-encrypt the post-match trio in PostMatchBalanceShare field order:
        // relayer_fee_balance (idx 5), protocol_fee_balance (6), amount (7)
        Fr p0 = nb.share.next(), p1 = nb.share.next(), p2 = nb.share.next();
        v.post_match_balance_shares = {nb.inner.relayer_fee_balance.sub(p0),
                                       nb.inner.protocol_fee_balance.sub(p1),
                                       nb.inner.amount.sub(p2)};
        nb.public_share[5] = v.post_match_balance_shares.relayer_fee_balance;
        nb.public_share[6] = v.post_match_balance_shares.protocol_fee_balance;
        nb.public_share[7] = v.post_match_balance_shares.amount;
        std::vector<Fr> new_priv_b = old_priv_b;
        new_priv_b[5] = p0;
        new_priv_b[6] = p1;
        new_priv_b[7] = p2;
        st.balance_recovery_id = nb.recovery.next();
        native_partial_commitment(
            new_priv_b, nb.recovery, nb.share,
            std::vector<Fr>(nb.public_share, nb.public_share + 8),
            BALANCE_PARTIAL_COMMITMENT_SIZE, st.balance_partial_private,
            st.balance_partial_public);

        // feed the re-encrypted shares back into the settlement witness: the
        // validity circuit's outputs ARE the settlement's pre-update shares
        b.sw.p[i].pre_amount_share = v.new_amount_public_share;
        b.sw.p[i].pre_in_shares = v.post_match_balance_shares;

        // --- OUTPUT BALANCE VALIDITY for this party (output_balance.rs) ---
        ObValidityWitness& o = b.ow[i];
        ObValidityStatement& os = b.ost[i];
        o.old_balance.inner = b.sw.p[i].output_balance;
        o.old_balance.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
        o.old_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
        for (int k = 0; k < 8; ++k) o.old_balance.public_share[k] = rng.fr();
        for (int k = 0; k < MERKLE_HEIGHT; ++k) {
            o.opening_elems[k] = rng.fr();
            o.opening_idx[k] = rng.next() & 1;
        }
        o.balance = b.sw.p[i].output_balance;
        auto old_ov = o.old_balance.inner.to_scalars();
        std::vector<Fr> old_priv_o = priv_of(old_ov, o.old_balance.public_share, 8);
        Fr old_comm_o = native_commitment(
            old_priv_o, o.old_balance.recovery, o.old_balance.share,
            std::vector<Fr>(o.old_balance.public_share, o.old_balance.public_share + 8));
        os.merkle_root = native_merkle_root(
            old_comm_o,
            std::vector<Fr>(o.opening_elems, o.opening_elems + MERKLE_HEIGHT),
            std::vector<bool>(o.opening_idx, o.opening_idx + MERKLE_HEIGHT));
        os.old_balance_nullifier = native_nullifier(o.old_balance.recovery);
        StateBalance no = o.old_balance;
        Fr q0 = no.share.next(), q1 = no.share.next(), q2 = no.share.next();
        o.post_match_balance_shares = {no.inner.relayer_fee_balance.sub(q0),
                                       no.inner.protocol_fee_balance.sub(q1),
                                       no.inner.amount.sub(q2)};
        no.public_share[5] = o.post_match_balance_shares.relayer_fee_balance;
        no.public_share[6] = o.post_match_balance_shares.protocol_fee_balance;
        no.public_share[7] = o.post_match_balance_shares.amount;
        std::vector<Fr> new_priv_o = old_priv_o;
        new_priv_o[5] = q0;
        new_priv_o[6] = q1;
        new_priv_o[7] = q2;
        os.recovery_id = no.recovery.next();
        native_partial_commitment(
            new_priv_o, no.recovery, no.share,
            std::vector<Fr>(no.public_share, no.public_share + 8),
            BALANCE_PARTIAL_COMMITMENT_SIZE, os.partial_private, os.partial_public);
        b.sw.p[i].pre_out_shares = o.post_match_balance_shares;
    }
    // recompute the statement fields that depend on the pre-update shares
    settlement_update_statement(b.sw, b.sst);
}

// bounded statement for bundle party 0 (worst-case bounds consistent with
// the bundle's obligation: max_in = obligation.amount_in, price = the trade
// price = 2 * intent.min_price)
inline void ib_bounded_statement_from_bundle(const ValidityBundle& b,
                                             IbBoundedStatement& st) {
    const auto& p = b.sw.p[0];
    st.bmr.internal_party_input_token = p.intent.in_token;
    st.bmr.internal_party_output_token = p.intent.out_token;
    st.bmr.max_internal_party_amount_in = p.obligation.amount_in;
    u64 l[4];
    p.obligation.amount_in.to_canonical(l);
    st.bmr.min_internal_party_amount_in = Fr::from_u64(l[0] / 2 + 1);
    st.bmr.price_repr = p.intent.min_price_repr.dbl();
    st.bmr.block_deadline = Fr::from_u64(123456);
    st.amount_public_share = p.pre_amount_share;
    st.in_shares = p.pre_in_shares;
    st.out_shares = p.pre_out_shares;
    st.internal_relayer_fee_repr = b.sst.relayer_fee_repr[0];
    st.external_relayer_fee_repr = b.sst.relayer_fee_repr[1];
    st.relayer_fee_recipient = p.output_balance.relayer_fee_recipient;
}

// statement for bundle party 0 (test_helpers :263-303 semantics)
inline void pub_settlement_statement_from_bundle(const ValidityBundle& b,
                                                 PubSettlementStatement& st) {
    st.obligation = b.sw.p[0].obligation;
    st.amount_public_share = b.sw.p[0].pre_amount_share;
    st.in_shares = b.sw.p[0].pre_in_shares;
    st.out_shares = b.sw.p[0].pre_out_shares;
    st.relayer_fee_repr = b.sst.relayer_fee_repr[0];
    st.protocol_fee_repr = b.sst.protocol_fee_repr;
    st.relayer_fee_recipient = b.sw.p[0].output_balance.relayer_fee_recipient;
}

// apply_constraints (output_balance.rs:62-130); link groups placed at the
// settlement's output_balance layout.
inline void ob_validity_apply_constraints(PlonkCircuit& cs, const ObValidityWitness& w,
                                          const ObValidityStatement& st, int alignment,
                                          int64_t party_offset0, int64_t party_offset1) {
    const char* g0 = "output_balance_settlement_party0";
    const char* g1 = "output_balance_settlement_party1";
    cs.create_link_group(g0, alignment, party_offset0);
    cs.create_link_group(g1, alignment, party_offset1);
    auto link_both = [&](Var x) {
        cs.add_to_link_group(x, g0);
        cs.add_to_link_group(x, g1);
    };

    // --- witness allocation (struct field order) ---
    StateWrapperVars ob_;
    ob_.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                    cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    ob_.share = {cs.create_variable(w.old_balance.share.seed),
                 cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    for (auto& s : w.old_balance.inner.to_scalars())
        ob_.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 8; ++k)
        ob_.public_share.push_back(cs.create_variable(w.old_balance.public_share[k]));
    std::vector<Var> op_elems, op_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_elems.push_back(cs.create_variable(w.opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_idx[k] ? Fr::one() : Fr::zero()));
    std::array<Var, 8> bal_v;
    {
        auto bv = w.balance.to_scalars();
        for (int k = 0; k < 8; ++k) {
            bal_v[k] = cs.create_variable(bv[k]);
            link_both(bal_v[k]);
        }
    }
    std::array<Var, 3> pms_v;
    {
        auto pv = w.post_match_balance_shares.to_scalars();
        for (int k = 0; k < 3; ++k) {
            pms_v[k] = cs.create_variable(pv[k]);
            link_both(pms_v[k]);
        }
    }

    // --- statement (public inputs, field order) ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var p_root = pub[0], p_null = pub[1], p_priv = pub[2], p_pub = pub[3],
        p_rid = pub[4];

    // complementary private shares (:70-75)
    std::vector<Var> old_priv;
    for (int k = 0; k < 8; ++k)
        old_priv.push_back(cs.sub(ob_.inner[k], ob_.public_share[k]));
    // build_new_balance (:104-130): re-encrypt the post-match trio
    CsprngVar nb_share = ob_.share;
    std::vector<Var> pads, cipher;
    stream_cipher_encrypt(cs, {ob_.inner[5], ob_.inner[6], ob_.inner[7]}, nb_share,
                          pads, cipher);
    std::vector<Var> new_priv = old_priv;
    new_priv[5] = pads[0];
    new_priv[6] = pads[1];
    new_priv[7] = pads[2];
    std::vector<Var> new_pub = ob_.public_share;
    new_pub[5] = cipher[0];
    new_pub[6] = cipher[1];
    new_pub[7] = cipher[2];
    for (int k = 0; k < 3; ++k) cs.enforce_equal(cipher[k], pms_v[k]);
    for (int k = 0; k < 8; ++k) cs.enforce_equal(ob_.inner[k], bal_v[k]);
    // rotation with partial commitment (state_rotation.rs:130-168)
    CsprngVar nb_rec = ob_.recovery;
    Var rid = csprng_next(cs, nb_rec);
    cs.enforce_equal(rid, p_rid);
    auto pc = partial_commitment_gadget(cs, new_priv, nb_rec, nb_share, new_pub,
                                        BALANCE_PARTIAL_COMMITMENT_SIZE);
    cs.enforce_equal(pc.first, p_priv);
    cs.enforce_equal(pc.second, p_pub);
    Var old_comm = commitment_gadget(cs, old_priv, ob_.recovery, ob_.share,
                                     ob_.public_share);
    Var root = merkle_root_gadget(cs, old_comm, op_elems, op_idx);
    cs.enforce_equal(root, p_root);
    Var nul = nullifier_gadget(cs, ob_);
    cs.enforce_equal(nul, p_null);
}

// apply_constraints (intent_and_balance.rs:71-232).  The two link groups are
// placed at the SETTLEMENT circuit's layout (proof_linking_groups inherits
// the settlement placement, :316-341): pass that placement in.
inline void validity_apply_constraints(PlonkCircuit& cs, const ValidityWitness& w,
                                       const ValidityStatement& st, int alignment,
                                       int64_t party_offset0, int64_t party_offset1) {
    const char* g0 = "intent_and_balance_settlement_party0";
    const char* g1 = "intent_and_balance_settlement_party1";
    cs.create_link_group(g0, alignment, party_offset0);
    cs.create_link_group(g1, alignment, party_offset1);
    auto link_both = [&](Var x) {
        cs.add_to_link_group(x, g0);
        cs.add_to_link_group(x, g1);
    };

    // --- witness allocation (struct field order) ---
    StateWrapperVars oi;
    oi.recovery = {cs.create_variable(w.old_intent.recovery.seed),
                   cs.create_variable(Fr::from_u64(w.old_intent.recovery.index))};
    oi.share = {cs.create_variable(w.old_intent.share.seed),
                cs.create_variable(Fr::from_u64(w.old_intent.share.index))};
    for (auto& s : w.old_intent.inner.to_scalars())
        oi.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 5; ++k)
        oi.public_share.push_back(cs.create_variable(w.old_intent.public_share[k]));
    std::vector<Var> iop_elems, iop_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        iop_elems.push_back(cs.create_variable(w.intent_opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        iop_idx.push_back(cs.create_boolean_variable(
            w.intent_opening_idx[k] ? Fr::one() : Fr::zero()));
    std::array<Var, 5> intent_v;
    {
        auto iv = w.intent.to_scalars();
        for (int k = 0; k < 5; ++k) {
            intent_v[k] = cs.create_variable(iv[k]);
            link_both(intent_v[k]);
        }
    }
    Var new_amt_share = cs.create_variable(w.new_amount_public_share);
    link_both(new_amt_share);
    StateWrapperVars ob_;
    ob_.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                    cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    ob_.share = {cs.create_variable(w.old_balance.share.seed),
                 cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    for (auto& s : w.old_balance.inner.to_scalars())
        ob_.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 8; ++k)
        ob_.public_share.push_back(cs.create_variable(w.old_balance.public_share[k]));
    std::vector<Var> bop_elems, bop_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        bop_elems.push_back(cs.create_variable(w.balance_opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        bop_idx.push_back(cs.create_boolean_variable(
            w.balance_opening_idx[k] ? Fr::one() : Fr::zero()));
    std::array<Var, 8> bal_v;
    {
        auto bv = w.balance.to_scalars();
        for (int k = 0; k < 8; ++k) {
            bal_v[k] = cs.create_variable(bv[k]);
            link_both(bal_v[k]);
        }
    }
    std::array<Var, 3> pms_v;
    {
        auto pv = w.post_match_balance_shares.to_scalars();
        for (int k = 0; k < 3; ++k) {
            pms_v[k] = cs.create_variable(pv[k]);
            link_both(pms_v[k]);
        }
    }

    // --- statement (public inputs, field order) ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var p_iroot = pub[0], p_inull = pub[1], p_ipriv = pub[2], p_ipub = pub[3],
        p_irid = pub[4];
    Var p_broot = pub[5], p_bnull = pub[6], p_bpriv = pub[7], p_bpub = pub[8],
        p_brid = pub[9];

    // --- validate_intent (:95-131) ---
    for (int k = 0; k < 5; ++k) cs.enforce_equal(intent_v[k], oi.inner[k]);
    std::vector<Var> old_priv_i;
    for (int k = 0; k < 5; ++k)
        old_priv_i.push_back(cs.sub(oi.inner[k], oi.public_share[k]));
    // build_new_intent (:135-158): re-encrypt amount_in (last share)
    CsprngVar ni_share = oi.share;
    std::vector<Var> ipads, icipher;
    stream_cipher_encrypt(cs, {oi.inner[4]}, ni_share, ipads, icipher);
    std::vector<Var> new_priv_i = old_priv_i;
    new_priv_i[4] = ipads[0];
    std::vector<Var> new_pub_i = oi.public_share;
    new_pub_i[4] = icipher[0];
    cs.enforce_equal(icipher[0], new_amt_share);
    // rotate_version_with_partial_commitment (state_rotation.rs:130-168)
    CsprngVar ni_rec = oi.recovery;
    Var irid = csprng_next(cs, ni_rec);
    cs.enforce_equal(irid, p_irid);
    auto ipc = partial_commitment_gadget(cs, new_priv_i, ni_rec, ni_share, new_pub_i,
                                         INTENT_PARTIAL_COMMITMENT_SIZE);
    cs.enforce_equal(ipc.first, p_ipriv);
    cs.enforce_equal(ipc.second, p_ipub);
    Var old_comm_i = commitment_gadget(cs, old_priv_i, oi.recovery, oi.share,
                                       oi.public_share);
    Var iroot = merkle_root_gadget(cs, old_comm_i, iop_elems, iop_idx);
    cs.enforce_equal(iroot, p_iroot);
    Var inull = nullifier_gadget(cs, oi);
    cs.enforce_equal(inull, p_inull);

    // --- validate_balance (:160-196) ---
    for (int k = 0; k < 8; ++k) cs.enforce_equal(bal_v[k], ob_.inner[k]);
    std::vector<Var> old_priv_b;
    for (int k = 0; k < 8; ++k)
        old_priv_b.push_back(cs.sub(ob_.inner[k], ob_.public_share[k]));
    // build_new_balance (:199-238): re-encrypt the post-match trio in
    // PostMatchBalanceShare field order (rfb@5, pfb@6, amount@7)
    CsprngVar nb_share = ob_.share;
    std::vector<Var> bpads, bcipher;
    stream_cipher_encrypt(cs, {ob_.inner[5], ob_.inner[6], ob_.inner[7]}, nb_share,
                          bpads, bcipher);
    std::vector<Var> new_priv_b = old_priv_b;
    new_priv_b[5] = bpads[0];
    new_priv_b[6] = bpads[1];
    new_priv_b[7] = bpads[2];
    std::vector<Var> new_pub_b = ob_.public_share;
    new_pub_b[5] = bcipher[0];
    new_pub_b[6] = bcipher[1];
    new_pub_b[7] = bcipher[2];
    for (int k = 0; k < 3; ++k) cs.enforce_equal(bcipher[k], pms_v[k]);
    CsprngVar nb_rec = ob_.recovery;
    Var brid = csprng_next(cs, nb_rec);
    cs.enforce_equal(brid, p_brid);
    auto bpc = partial_commitment_gadget(cs, new_priv_b, nb_rec, nb_share, new_pub_b,
                                         BALANCE_PARTIAL_COMMITMENT_SIZE);
    cs.enforce_equal(bpc.first, p_bpriv);
    cs.enforce_equal(bpc.second, p_bpub);
    Var old_comm_b = commitment_gadget(cs, old_priv_b, ob_.recovery, ob_.share,
                                       ob_.public_share);
    Var broot = merkle_root_gadget(cs, old_comm_b, bop_elems, bop_idx);
    cs.enforce_equal(broot, p_broot);
    Var bnull = nullifier_gadget(cs, ob_);
    cs.enforce_equal(bnull, p_bnull);

    // --- intent <-> balance cross constraints (:84-88) ---
    cs.enforce_equal(intent_v[0], bal_v[0]);  // in_token == mint
    cs.enforce_equal(intent_v[2], bal_v[1]);  // owner == owner
}

// ========== INTENT AND BALANCE FIRST FILL VALIDITY ==========
// (validity_proofs/intent_and_balance_first_fill.rs — first fill: the intent
//  is NOT yet in the tree; its authorization is bootstrapped from the
//  balance's authority key via a Schnorr signature over the intent's
//  commitment.)

struct FfWitness {  // :285-329 (field order)
    Intent intent;  // linked party0+party1
    Csprng share_stream, recovery_stream;
    Fr private_intent_shares[5];
    Fr new_amount_public_share;  // linked
    JjSignature sig;             // SchnorrSignature {s, R}
    StateBalance old_balance;
    Balance balance;                           // linked
    PostMatchShare post_match_balance_shares;  // linked
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_idx[MERKLE_HEIGHT];
};
struct FfStatement {  // :341-368 (11 scalars)
    Fr merkle_root;
    Fr intent_public_share[4];  // PreMatchIntentShare (no amount_in)
    Fr intent_private_share_commitment, intent_recovery_id;
    Fr balance_partial_private, balance_partial_public;
    Fr old_balance_nullifier, balance_recovery_id;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = {merkle_root};
        v.insert(v.end(), intent_public_share, intent_public_share + 4);
        v.push_back(intent_private_share_commitment);
        v.push_back(intent_recovery_id);
        v.push_back(balance_partial_private);
        v.push_back(balance_partial_public);
        v.push_back(old_balance_nullifier);
        v.push_back(balance_recovery_id);
        return v;
    }
};

// First-fill witness for bundle party p: shares the intent, balance state and
// re-encrypted shares with the bundle's intent_and_balance validity witness,
// so the first-fill proof links into the same settlement proof.
inline void ff_build(const ValidityBundle& b, int party, uint64_t seed, FfWitness& w,
                     FfStatement& st) {
    const ValidityWitness& v = b.vw[party];
    const ValidityStatement& vs = b.vst[party];
    Lcg rng(seed ^ 0xFF00FF00FF00FF0ull);
    w.intent = b.sw.p[party].intent;
    w.share_stream = {rng.fr(), rng.next() & 0xFFFF};
    w.recovery_stream = {rng.fr(), rng.next() & 0xFFFF};
    auto iv = w.intent.to_scalars();
    for (int k = 0; k < 4; ++k) w.private_intent_shares[k] = rng.fr();
    // arrange the fresh amount_in public share to equal the bundle's
    // (the settlement's pre-update share)
    w.new_amount_public_share = v.new_amount_public_share;
    w.private_intent_shares[4] = iv[4].sub(w.new_amount_public_share);
    w.old_balance = v.old_balance;
    w.balance = v.balance;
    w.post_match_balance_shares = v.post_match_balance_shares;
    for (int k = 0; k < MERKLE_HEIGHT; ++k) {
        w.opening_elems[k] = v.balance_opening_elems[k];
        w.opening_idx[k] = v.balance_opening_idx[k];
    }
    // public shares + commitments
    Fr pub[5];
    for (int k = 0; k < 5; ++k) pub[k] = iv[k].sub(w.private_intent_shares[k]);
    std::vector<Fr> priv(w.private_intent_shares, w.private_intent_shares + 5);
    auto priv_comm_at = [&](uint64_t rec_index) {
        std::vector<Fr> in = priv;
        in.push_back(w.recovery_stream.seed);
        in.push_back(Fr::from_u64(rec_index));
        in.push_back(w.share_stream.seed);
        in.push_back(Fr::from_u64(w.share_stream.index));
        return poseidon_hash(in.data(), in.size());
    };
    Fr old_pc = priv_comm_at(w.recovery_stream.index);
    Fr new_pc = priv_comm_at(w.recovery_stream.index + 1);
    Fr pub_comm = pub[0];
    for (int k = 1; k < 5; ++k) {
        Fr two[2] = {pub_comm, pub[k]};
        pub_comm = poseidon_hash(two, 2);
    }
    Fr two[2] = {old_pc, pub_comm};
    Fr full_comm = poseidon_hash(two, 2);
    Csprng rec = w.recovery_stream;
    st.intent_recovery_id = rec.next();
    st.intent_private_share_commitment = new_pc;
    for (int k = 0; k < 4; ++k) st.intent_public_share[k] = pub[k];
    st.merkle_root = vs.balance_merkle_root;
    st.balance_partial_private = vs.balance_partial_private;
    st.balance_partial_public = vs.balance_partial_public;
    st.old_balance_nullifier = vs.old_balance_nullifier;
    st.balance_recovery_id = vs.balance_recovery_id;
    // sign the ORIGINAL intent commitment with the balance's authority key
    JjScalar k_nonce = jj_random_scalar(rng);
    w.sig = jj_sign(b.auth_sk[party], k_nonce, &full_comm, 1);
}

inline void ff_apply_constraints(PlonkCircuit& cs, const FfWitness& w,
                                 const FfStatement& st, int alignment,
                                 int64_t party_offset0, int64_t party_offset1) {
    const char* g0 = "intent_and_balance_settlement_party0";
    const char* g1 = "intent_and_balance_settlement_party1";
    cs.create_link_group(g0, alignment, party_offset0);
    cs.create_link_group(g1, alignment, party_offset1);
    auto link_both = [&](Var x) {
        cs.add_to_link_group(x, g0);
        cs.add_to_link_group(x, g1);
    };

    // --- witness allocation (struct field order) ---
    std::array<Var, 5> intent_v;
    {
        auto iv = w.intent.to_scalars();
        for (int k = 0; k < 5; ++k) {
            intent_v[k] = cs.create_variable(iv[k]);
            link_both(intent_v[k]);
        }
    }
    CsprngVar share{cs.create_variable(w.share_stream.seed),
                    cs.create_variable(Fr::from_u64(w.share_stream.index))};
    CsprngVar rec{cs.create_variable(w.recovery_stream.seed),
                  cs.create_variable(Fr::from_u64(w.recovery_stream.index))};
    std::vector<Var> ipriv;
    for (int k = 0; k < 5; ++k)
        ipriv.push_back(cs.create_variable(w.private_intent_shares[k]));
    Var new_amt = cs.create_variable(w.new_amount_public_share);
    link_both(new_amt);
    Var sig_s = cs.create_variable(Fr::from_canonical(w.sig.s.v));
    JjPointVars sig_R{cs.create_variable(w.sig.R.x), cs.create_variable(w.sig.R.y)};
    StateWrapperVars ob_;
    ob_.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                    cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    ob_.share = {cs.create_variable(w.old_balance.share.seed),
                 cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    for (auto& s : w.old_balance.inner.to_scalars())
        ob_.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 8; ++k)
        ob_.public_share.push_back(cs.create_variable(w.old_balance.public_share[k]));
    std::array<Var, 8> bal_v;
    {
        auto bv = w.balance.to_scalars();
        for (int k = 0; k < 8; ++k) {
            bal_v[k] = cs.create_variable(bv[k]);
            link_both(bal_v[k]);
        }
    }
    std::array<Var, 3> pms_v;
    {
        auto pv = w.post_match_balance_shares.to_scalars();
        for (int k = 0; k < 3; ++k) {
            pms_v[k] = cs.create_variable(pv[k]);
            link_both(pms_v[k]);
        }
    }
    std::vector<Var> op_elems, op_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_elems.push_back(cs.create_variable(w.opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_idx[k] ? Fr::one() : Fr::zero()));

    // --- statement ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var p_root = pub[0];
    std::array<Var, 4> p_ipub{pub[1], pub[2], pub[3], pub[4]};
    Var p_ipc = pub[5], p_irid = pub[6];
    Var p_bpriv = pub[7], p_bpub = pub[8], p_bnull = pub[9], p_brid = pub[10];

    // --- validate_balance (:218-238 + create_new_balance :240-277) ---
    for (int k = 0; k < 8; ++k) cs.enforce_equal(bal_v[k], ob_.inner[k]);
    std::vector<Var> old_priv_b;
    for (int k = 0; k < 8; ++k)
        old_priv_b.push_back(cs.sub(ob_.inner[k], ob_.public_share[k]));
    CsprngVar nb_share = ob_.share;
    std::vector<Var> bpads, bcipher;
    stream_cipher_encrypt(cs, {ob_.inner[5], ob_.inner[6], ob_.inner[7]}, nb_share,
                          bpads, bcipher);
    std::vector<Var> new_priv_b = old_priv_b;
    new_priv_b[5] = bpads[0];
    new_priv_b[6] = bpads[1];
    new_priv_b[7] = bpads[2];
    std::vector<Var> new_pub_b = ob_.public_share;
    new_pub_b[5] = bcipher[0];
    new_pub_b[6] = bcipher[1];
    new_pub_b[7] = bcipher[2];
    for (int k = 0; k < 3; ++k) cs.enforce_equal(bcipher[k], pms_v[k]);
    CsprngVar nb_rec = ob_.recovery;
    Var brid = csprng_next(cs, nb_rec);
    cs.enforce_equal(brid, p_brid);
    auto bpc = partial_commitment_gadget(cs, new_priv_b, nb_rec, nb_share, new_pub_b,
                                         BALANCE_PARTIAL_COMMITMENT_SIZE);
    cs.enforce_equal(bpc.first, p_bpriv);
    cs.enforce_equal(bpc.second, p_bpub);
    Var old_comm_b = commitment_gadget(cs, old_priv_b, ob_.recovery, ob_.share,
                                       ob_.public_share);
    Var broot = merkle_root_gadget(cs, old_comm_b, op_elems, op_idx);
    cs.enforce_equal(broot, p_root);
    Var bnull = nullifier_gadget(cs, ob_);
    cs.enforce_equal(bnull, p_bnull);

    // --- verify_intent_fields (:120-140) ---
    cs.enforce_equal(intent_v[2], bal_v[1]);  // owner
    cs.enforce_in_range(intent_v[3], PRICE_BITS);
    cs.enforce_in_range(intent_v[4], AMOUNT_BITS);
    cs.enforce_equal(bal_v[0], intent_v[0]);  // balance.mint == intent.in_token
    // --- build_intent_state_wrapper (:142-169) ---
    std::vector<Var> ipub;
    for (int k = 0; k < 5; ++k) ipub.push_back(cs.sub(intent_v[k], ipriv[k]));
    for (int k = 0; k < 4; ++k) cs.enforce_equal(ipub[k], p_ipub[k]);
    cs.enforce_equal(ipub[4], new_amt);
    // --- recovery id + commitments (:106-116, :171-203) ---
    CsprngVar ni_rec = rec;
    Var irid = csprng_next(cs, ni_rec);
    cs.enforce_equal(irid, p_irid);
    auto priv_comm_at = [&](const CsprngVar& r) {
        PoseidonHashGadget h(cs);
        std::vector<Var> in = ipriv;
        in.push_back(r.seed);
        in.push_back(r.index);
        in.push_back(share.seed);
        in.push_back(share.index);
        return h.hash(cs, in);
    };
    Var old_pc = priv_comm_at(rec);
    Var new_pc = priv_comm_at(ni_rec);
    cs.enforce_equal(new_pc, p_ipc);
    Var pub_comm = resumable_commitment(cs, ipub);
    PoseidonHashGadget hf(cs);
    Var full_comm = hf.hash(cs, {old_pc, pub_comm});
    // --- Schnorr authorization by the balance's authority (:79-86) ---
    JjPointVars vk{bal_v[3], bal_v[4]};
    schnorr_verify_gadget(cs, vk, sig_R, sig_s, {full_comm});
}

// ========== NEW OUTPUT BALANCE VALIDITY ==========
// (validity_proofs/new_output_balance.rs — create a fresh output balance
//  (zero amount/fees), authorized by a Schnorr signature from an EXISTING
//  balance's authority; links at the settlement's output-balance layout.)

struct NobWitness {  // :246-269 (field order)
    StateBalance new_balance;
    Balance balance;                           // linked out0+out1
    PostMatchShare post_match_balance_shares;  // linked
    StateBalance existing_balance;
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_idx[MERKLE_HEIGHT];
    JjSignature sig;
};
struct NobStatement {  // :281-299 (10 scalars)
    Fr existing_merkle_root, existing_nullifier;
    Fr pre_match_shares[5];  // mint, owner, relayer_fee_recipient, authority x/y
    Fr partial_private, partial_public, recovery_id;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = {existing_merkle_root, existing_nullifier};
        v.insert(v.end(), pre_match_shares, pre_match_shares + 5);
        v.push_back(partial_private);
        v.push_back(partial_public);
        v.push_back(recovery_id);
        return v;
    }
};

inline void nob_build(uint64_t seed, NobWitness& w, NobStatement& st) {
    Lcg rng(seed ^ 0x0B0B0B0B0B0B0B0Bull);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    JjScalar sk = jj_random_scalar(rng);
    JjPoint vk = jj_pubkey(sk);
    // existing balance (the authorization bootstrap)
    w.existing_balance.inner = {addr(), addr(), addr(), vk.x, vk.y,
                                Fr::from_u64(rng.next() & 0xFFFF),
                                Fr::from_u64(rng.next() & 0xFFFF),
                                Fr::from_u64(rng.next() & ((1ull << 50) - 1))};
    w.existing_balance.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
    w.existing_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int k = 0; k < 8; ++k) w.existing_balance.public_share[k] = rng.fr();
    for (int k = 0; k < MERKLE_HEIGHT; ++k) {
        w.opening_elems[k] = rng.fr();
        w.opening_idx[k] = rng.next() & 1;
    }
    // fresh output balance: zero amount and fees, same owner/authority/rfr
    w.new_balance.inner = {addr(), w.existing_balance.inner.owner,
                           w.existing_balance.inner.relayer_fee_recipient, vk.x, vk.y,
                           Fr::zero(), Fr::zero(), Fr::zero()};
    w.new_balance.recovery = {rng.fr(), rng.next() & 0xFFFF};
    w.new_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int k = 0; k < 8; ++k) w.new_balance.public_share[k] = rng.fr();
    w.balance = w.new_balance.inner;
    w.post_match_balance_shares = {w.new_balance.public_share[5],
                                   w.new_balance.public_share[6],
                                   w.new_balance.public_share[7]};
    // natives
    auto priv_of = [](const Balance& inner, const Fr* pub_) {
        std::vector<Fr> p;
        auto iv = inner.to_scalars();
        for (int i = 0; i < 8; ++i) p.push_back(iv[i].sub(pub_[i]));
        return p;
    };
    std::vector<Fr> priv_n = priv_of(w.new_balance.inner, w.new_balance.public_share);
    Fr full_comm;
    {
        // full commitment at the PRE-advance recovery index
        std::vector<Fr> pubs(w.new_balance.public_share, w.new_balance.public_share + 8);
        full_comm = native_commitment(priv_n, w.new_balance.recovery, w.new_balance.share,
                                      pubs);
    }
    StateBalance nb = w.new_balance;
    st.recovery_id = nb.recovery.next();
    native_partial_commitment(
        priv_n, nb.recovery, nb.share,
        std::vector<Fr>(nb.public_share, nb.public_share + 8),
        BALANCE_PARTIAL_COMMITMENT_SIZE, st.partial_private, st.partial_public);
    for (int k = 0; k < 5; ++k) st.pre_match_shares[k] = w.new_balance.public_share[k];
    std::vector<Fr> priv_e = priv_of(w.existing_balance.inner,
                                     w.existing_balance.public_share);
    Fr ecomm = native_commitment(
        priv_e, w.existing_balance.recovery, w.existing_balance.share,
        std::vector<Fr>(w.existing_balance.public_share,
                        w.existing_balance.public_share + 8));
    st.existing_merkle_root = native_merkle_root(
        ecomm, std::vector<Fr>(w.opening_elems, w.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(w.opening_idx, w.opening_idx + MERKLE_HEIGHT));
    st.existing_nullifier = native_nullifier(w.existing_balance.recovery);
    JjScalar k_nonce = jj_random_scalar(rng);
    w.sig = jj_sign(sk, k_nonce, &full_comm, 1);
}

inline void nob_apply_constraints(PlonkCircuit& cs, const NobWitness& w,
                                  const NobStatement& st, int alignment,
                                  int64_t out_offset0, int64_t out_offset1) {
    const char* g0 = "output_balance_settlement_party0";
    const char* g1 = "output_balance_settlement_party1";
    cs.create_link_group(g0, alignment, out_offset0);
    cs.create_link_group(g1, alignment, out_offset1);
    auto link_both = [&](Var x) {
        cs.add_to_link_group(x, g0);
        cs.add_to_link_group(x, g1);
    };

    // --- witness allocation (struct field order) ---
    StateWrapperVars nb;
    nb.recovery = {cs.create_variable(w.new_balance.recovery.seed),
                   cs.create_variable(Fr::from_u64(w.new_balance.recovery.index))};
    nb.share = {cs.create_variable(w.new_balance.share.seed),
                cs.create_variable(Fr::from_u64(w.new_balance.share.index))};
    for (auto& s : w.new_balance.inner.to_scalars())
        nb.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 8; ++k)
        nb.public_share.push_back(cs.create_variable(w.new_balance.public_share[k]));
    std::array<Var, 8> bal_v;
    {
        auto bv = w.balance.to_scalars();
        for (int k = 0; k < 8; ++k) {
            bal_v[k] = cs.create_variable(bv[k]);
            link_both(bal_v[k]);
        }
    }
    std::array<Var, 3> pms_v;
    {
        auto pv = w.post_match_balance_shares.to_scalars();
        for (int k = 0; k < 3; ++k) {
            pms_v[k] = cs.create_variable(pv[k]);
            link_both(pms_v[k]);
        }
    }
    StateWrapperVars eb;
    eb.recovery = {cs.create_variable(w.existing_balance.recovery.seed),
                   cs.create_variable(Fr::from_u64(w.existing_balance.recovery.index))};
    eb.share = {cs.create_variable(w.existing_balance.share.seed),
                cs.create_variable(Fr::from_u64(w.existing_balance.share.index))};
    for (auto& s : w.existing_balance.inner.to_scalars())
        eb.inner.push_back(cs.create_variable(s));
    for (int k = 0; k < 8; ++k)
        eb.public_share.push_back(cs.create_variable(w.existing_balance.public_share[k]));
    std::vector<Var> op_elems, op_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_elems.push_back(cs.create_variable(w.opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_idx[k] ? Fr::one() : Fr::zero()));
    Var sig_s = cs.create_variable(Fr::from_canonical(w.sig.s.v));
    JjPointVars sig_R{cs.create_variable(w.sig.R.x), cs.create_variable(w.sig.R.y)};

    // --- statement ---
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    Var p_eroot = pub[0], p_enull = pub[1];
    std::array<Var, 5> p_pre{pub[2], pub[3], pub[4], pub[5], pub[6]};
    Var p_priv = pub[7], p_pub = pub[8], p_rid = pub[9];

    // --- validate_new_balance (:144-166) ---
    for (int k = 0; k < 8; ++k) cs.enforce_equal(bal_v[k], nb.inner[k]);
    cs.enforce_equal(bal_v[7], cs.zero());  // amount
    cs.enforce_equal(bal_v[5], cs.zero());  // relayer fee balance
    cs.enforce_equal(bal_v[6], cs.zero());  // protocol fee balance
    // --- validate_balance_shares (:116-141) ---
    std::vector<Var> priv_n;
    for (int k = 0; k < 8; ++k)
        priv_n.push_back(cs.sub(nb.inner[k], nb.public_share[k]));
    for (int k = 0; k < 5; ++k) cs.enforce_equal(nb.public_share[k], p_pre[k]);
    for (int k = 0; k < 3; ++k) cs.enforce_equal(nb.public_share[5 + k], pms_v[k]);
    // --- recovery id + shared-prefix commitments (:83-107) ---
    CsprngVar nb_rec2 = nb.recovery;
    Var rid = csprng_next(cs, nb_rec2);
    cs.enforce_equal(rid, p_rid);
    Var full_comm = commitment_gadget(cs, priv_n, nb.recovery, nb.share,
                                      nb.public_share);
    auto pc = partial_commitment_gadget(cs, priv_n, nb_rec2, nb.share, nb.public_share,
                                        BALANCE_PARTIAL_COMMITMENT_SIZE);
    cs.enforce_equal(pc.first, p_priv);
    cs.enforce_equal(pc.second, p_pub);
    // --- authorize_new_balance (:168-228) ---
    std::vector<Var> priv_e;
    for (int k = 0; k < 8; ++k)
        priv_e.push_back(cs.sub(eb.inner[k], eb.public_share[k]));
    Var ecomm = commitment_gadget(cs, priv_e, eb.recovery, eb.share, eb.public_share);
    Var eroot = merkle_root_gadget(cs, ecomm, op_elems, op_idx);
    cs.enforce_equal(eroot, p_eroot);
    Var enull = nullifier_gadget(cs, eb);
    cs.enforce_equal(enull, p_enull);
    JjPointVars vk{eb.inner[3], eb.inner[4]};
    schnorr_verify_gadget(cs, vk, sig_R, sig_s, {full_comm});
    cs.enforce_equal(bal_v[1], eb.inner[1]);  // owner
    cs.enforce_equal(bal_v[3], eb.inner[3]);  // authority
    cs.enforce_equal(bal_v[4], eb.inner[4]);
    cs.enforce_equal(bal_v[2], eb.inner[2]);  // relayer fee recipient
}

// ================== Fee payment circuits (zk_circuits/fees/) ==================
// Notes are plaintext 4-tuples committed with Poseidon2; fee payments rotate
// the payer's balance with the fee field re-encrypted to zero.

struct Note {  // darkpool-types/src/note.rs:47-59 (4 scalars)
    Fr mint, amount, receiver, blinder;
    std::vector<Fr> to_scalars() const { return {mint, amount, receiver, blinder}; }
};
// note.rs:77-87: commitment = H(fields); nullifier = H(commitment, blinder)
inline Fr native_note_commitment(const Note& n) {
    Fr in[4] = {n.mint, n.amount, n.receiver, n.blinder};
    return poseidon_hash(in, 4);
}
inline Fr native_note_nullifier(const Fr& comm, const Fr& blinder) {
    Fr in[2] = {comm, blinder};
    return poseidon_hash(in, 2);
}

// ---- VALID NOTE REDEMPTION (fees/valid_note_redemption.rs) ----
struct NoteRedemptionStatement {  // :88-97 (6 scalars)
    Note note;
    Fr note_root, note_nullifier;
    std::vector<Fr> to_scalars() const {
        std::vector<Fr> v = note.to_scalars();
        v.push_back(note_root);
        v.push_back(note_nullifier);
        return v;
    }
};
struct NoteRedemptionWitness {  // :73-77
    Fr opening_elems[MERKLE_HEIGHT];
    bool opening_idx[MERKLE_HEIGHT];
};

inline void note_redemption_build(uint64_t seed, NoteRedemptionWitness& w,
                                  NoteRedemptionStatement& st) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    st.note = {addr(), Fr::from_u64(rng.next() & ((1ull << 50) - 1)), addr(), rng.fr()};
    for (int k = 0; k < MERKLE_HEIGHT; ++k) {
        w.opening_elems[k] = rng.fr();
        w.opening_idx[k] = rng.next() & 1;
    }
    Fr comm = native_note_commitment(st.note);
    st.note_root = native_merkle_root(
        comm, std::vector<Fr>(w.opening_elems, w.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(w.opening_idx, w.opening_idx + MERKLE_HEIGHT));
    st.note_nullifier = native_note_nullifier(comm, st.note.blinder);
}

inline void note_redemption_apply_constraints(PlonkCircuit& cs,
                                              const NoteRedemptionWitness& w,
                                              const NoteRedemptionStatement& st) {
    std::vector<Var> op_elems, op_idx;
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_elems.push_back(cs.create_variable(w.opening_elems[k]));
    for (int k = 0; k < MERKLE_HEIGHT; ++k)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_idx[k] ? Fr::one() : Fr::zero()));
    auto ss = st.to_scalars();
    std::vector<Var> pub;
    for (auto& s : ss) pub.push_back(cs.create_public_variable(s));
    // NoteGadget::compute_note_commitment (note hash over the 4 fields)
    PoseidonHashGadget h(cs);
    Var comm = h.hash(cs, {pub[0], pub[1], pub[2], pub[3]});
    Var root = merkle_root_gadget(cs, comm, op_elems, op_idx);
    cs.enforce_equal(root, pub[4]);
    PoseidonHashGadget h2(cs);
    Var nul = h2.hash(cs, {comm, pub[3]});
    cs.enforce_equal(nul, pub[5]);
}

// ---- VALID {PUBLIC, PRIVATE} x {RELAYER, PROTOCOL} FEE PAYMENT ----
// (fees/valid_public_relayer_fee_payment.rs etc.) — shared machinery: the
// payer balance rotates with fee field `field` (5 = relayer, 6 = protocol)
// re-encrypted to zero; public variants carry the full plaintext note in the
// statement, the private relayer variant only the receiver + note commitment
// (encryption verified out of circuit).

struct FeePaymentStatement {  // balance-rotation half (5 scalars), shared
    Fr merkle_root, old_balance_nullifier, new_balance_commitment, recovery_id;
    Fr new_fee_balance_share;
};

// builder: balance with nonzero fee balance at `field`, full rotation natives
inline void fee_payment_build(uint64_t seed, int field, VdWitness& w,
                              FeePaymentStatement& st, Note& note) {
    Lcg rng(seed);
    auto addr = [&]() {
        u64 l[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                    rng.next() & 0xFFFFFFFF, 0};
        return Fr::from_canonical(l);
    };
    w.old_balance.inner = {addr(), addr(), addr(), rng.fr(), rng.fr(),
                           Fr::from_u64((rng.next() & 0xFFFF) + 1),
                           Fr::from_u64((rng.next() & 0xFFFF) + 1),
                           Fr::from_u64(rng.next() & ((1ull << 50) - 1))};
    w.old_balance.recovery = {rng.fr(), (rng.next() & 0xFFFF) + 1};
    w.old_balance.share = {rng.fr(), rng.next() & 0xFFFFFF};
    for (int i = 0; i < 8; ++i) w.old_balance.public_share[i] = rng.fr();
    for (int i = 0; i < MERKLE_HEIGHT; ++i) {
        w.opening_elems[i] = rng.fr();
        w.opening_indices[i] = rng.next() & 1;
    }
    auto iv = w.old_balance.inner.to_scalars();
    std::vector<Fr> old_priv;
    for (int i = 0; i < 8; ++i)
        old_priv.push_back(iv[i].sub(w.old_balance.public_share[i]));
    Fr old_comm = native_commitment(
        old_priv, w.old_balance.recovery, w.old_balance.share,
        std::vector<Fr>(w.old_balance.public_share, w.old_balance.public_share + 8));
    st.merkle_root = native_merkle_root(
        old_comm, std::vector<Fr>(w.opening_elems, w.opening_elems + MERKLE_HEIGHT),
        std::vector<bool>(w.opening_indices, w.opening_indices + MERKLE_HEIGHT));
    st.old_balance_nullifier = native_nullifier(w.old_balance.recovery);
    // the note pays out the full fee balance to the recipient on the balance
    note.mint = w.old_balance.inner.mint;
    note.amount = iv[field];
    note.receiver = w.old_balance.inner.relayer_fee_recipient;
    note.blinder = rng.fr();
    // rotate with fee field := 0, re-encrypted
    StateBalance nb = w.old_balance;
    Fr pad = nb.share.next();
    st.new_fee_balance_share = Fr::zero().sub(pad);
    nb.public_share[field] = st.new_fee_balance_share;
    std::vector<Fr> new_priv = old_priv;
    new_priv[field] = pad;
    if (field == 5) nb.inner.relayer_fee_balance = Fr::zero();
    else nb.inner.protocol_fee_balance = Fr::zero();
    st.recovery_id = nb.recovery.next();
    auto nv2 = nb.inner.to_scalars();
    std::vector<Fr> new_pub(nb.public_share, nb.public_share + 8);
    st.new_balance_commitment = native_commitment(new_priv, nb.recovery, nb.share, new_pub);
}

// shared constraint body; note_mode: 0 = public note (9 statement scalars:
// rotation 5 + note 4), 1 = private note (7: rotation 5 + receiver +
// note_commitment); check_receiver: constrain note.receiver (false only for
// the public protocol variant, whose receiver the contract checks)
inline void fee_payment_apply_constraints(PlonkCircuit& cs, const VdWitness& w,
                                          const Fr& blinder, int field, int note_mode,
                                          bool check_receiver,
                                          const std::vector<Fr>& st_scalars) {
    StateWrapperVars old_v;
    old_v.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                      cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    old_v.share = {cs.create_variable(w.old_balance.share.seed),
                   cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    auto iv = w.old_balance.inner.to_scalars();
    for (auto& s : iv) old_v.inner.push_back(cs.create_variable(s));
    for (int i = 0; i < 8; ++i)
        old_v.public_share.push_back(cs.create_variable(w.old_balance.public_share[i]));
    std::vector<Var> op_elems, op_idx;
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_elems.push_back(cs.create_variable(w.opening_elems[i]));
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_indices[i] ? Fr::one() : Fr::zero()));
    Var blinder_v = (note_mode == 1) ? cs.create_variable(blinder) : cs.zero();

    std::vector<Var> pub;
    for (auto& s : st_scalars) pub.push_back(cs.create_public_variable(s));
    Var p_root = pub[0], p_null = pub[1], p_comm = pub[2], p_rid = pub[3],
        p_fee_share = pub[4];

    // the fee balance must be nonzero (verify_note, :85-88)
    Var z = is_zero_gadget(cs, old_v.inner[field]);
    cs.enforce_false(z);
    // note checks
    if (note_mode == 0) {
        Var n_mint = pub[5], n_amount = pub[6], n_receiver = pub[7];
        cs.enforce_equal(n_mint, old_v.inner[0]);
        cs.enforce_equal(n_amount, old_v.inner[field]);
        if (check_receiver) cs.enforce_equal(n_receiver, old_v.inner[2]);
    } else {
        Var p_receiver = pub[5], p_note_comm = pub[6];
        cs.enforce_equal(p_receiver, old_v.inner[2]);
        PoseidonHashGadget h(cs);
        Var ncomm = h.hash(cs, {old_v.inner[0], old_v.inner[field], old_v.inner[2],
                                blinder_v});
        cs.enforce_equal(ncomm, p_note_comm);
    }

    // complementary shares + post-payment balance (fee field := 0)
    std::vector<Var> old_priv;
    for (int i = 0; i < 8; ++i)
        old_priv.push_back(cs.sub(old_v.inner[i], old_v.public_share[i]));
    StateWrapperVars new_v = old_v;
    std::vector<Var> new_priv = old_priv;
    new_v.inner[field] = cs.zero();
    std::vector<Var> pads, pubs_enc;
    stream_cipher_encrypt(cs, {new_v.inner[field]}, new_v.share, pads, pubs_enc);
    new_priv[field] = pads[0];
    new_v.public_share[field] = pubs_enc[0];
    cs.enforce_equal(pubs_enc[0], p_fee_share);

    // full rotation (state_rotation.rs:83-122)
    Var rid = csprng_next(cs, new_v.recovery);
    cs.enforce_equal(rid, p_rid);
    Var old_comm = commitment_gadget(cs, old_priv, old_v.recovery, old_v.share,
                                     old_v.public_share);
    Var new_comm = commitment_gadget(cs, new_priv, new_v.recovery, new_v.share,
                                     new_v.public_share);
    cs.enforce_equal(new_comm, p_comm);
    Var root = merkle_root_gadget(cs, old_comm, op_elems, op_idx);
    cs.enforce_equal(root, p_root);
    Var nul = nullifier_gadget(cs, old_v);
    cs.enforce_equal(nul, p_null);
}

// ---- VALID PRIVATE PROTOCOL FEE PAYMENT (ElGamal note encryption) ----
// (fees/valid_private_protocol_fee_payment.rs — the note is encrypted
//  IN-circuit under the protocol key; statement = rotation half + receiver +
//  note commitment + ciphertext (eph 2 + 3) + encryption key (2) = 14.)
inline void fee_private_protocol_apply_constraints(PlonkCircuit& cs, const VdWitness& w,
                                                   const Fr& blinder, const Fr& enc_k,
                                                   const std::vector<Fr>& st_scalars) {
    const int field = 6;  // protocol_fee_balance
    StateWrapperVars old_v;
    old_v.recovery = {cs.create_variable(w.old_balance.recovery.seed),
                      cs.create_variable(Fr::from_u64(w.old_balance.recovery.index))};
    old_v.share = {cs.create_variable(w.old_balance.share.seed),
                   cs.create_variable(Fr::from_u64(w.old_balance.share.index))};
    for (auto& s : w.old_balance.inner.to_scalars())
        old_v.inner.push_back(cs.create_variable(s));
    for (int i = 0; i < 8; ++i)
        old_v.public_share.push_back(cs.create_variable(w.old_balance.public_share[i]));
    std::vector<Var> op_elems, op_idx;
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_elems.push_back(cs.create_variable(w.opening_elems[i]));
    for (int i = 0; i < MERKLE_HEIGHT; ++i)
        op_idx.push_back(cs.create_boolean_variable(
            w.opening_indices[i] ? Fr::one() : Fr::zero()));
    Var blinder_v = cs.create_variable(blinder);
    Var k_v = cs.create_variable(enc_k);

    std::vector<Var> pub;
    for (auto& s : st_scalars) pub.push_back(cs.create_public_variable(s));
    Var p_root = pub[0], p_null = pub[1], p_comm = pub[2], p_rid = pub[3],
        p_fee_share = pub[4], p_receiver = pub[5], p_note_comm = pub[6];
    JjPointVars p_eph{pub[7], pub[8]};
    Var p_c0 = pub[9], p_c1 = pub[10], p_c2 = pub[11];
    JjPointVars p_pk{pub[12], pub[13]};

    // fee balance nonzero (:84-88)
    Var z = is_zero_gadget(cs, old_v.inner[field]);
    cs.enforce_false(z);
    // note = {mint, protocol_fee_balance, statement.receiver, blinder}
    // encryption: plaintext = [mint, amount, blinder] (note.rs:91-94)
    JjPointVars eph;
    std::vector<Var> cipher;
    elgamal_encrypt_gadget(cs, p_pk, k_v,
                           {old_v.inner[0], old_v.inner[field], blinder_v}, eph, cipher);
    cs.enforce_equal(eph.x, p_eph.x);
    cs.enforce_equal(eph.y, p_eph.y);
    cs.enforce_equal(cipher[0], p_c0);
    cs.enforce_equal(cipher[1], p_c1);
    cs.enforce_equal(cipher[2], p_c2);
    // note commitment (H over the 4 fields)
    PoseidonHashGadget hn(cs);
    Var ncomm = hn.hash(cs, {old_v.inner[0], old_v.inner[field], p_receiver, blinder_v});
    cs.enforce_equal(ncomm, p_note_comm);

    // post-payment rotation (protocol fee := 0)
    std::vector<Var> old_priv;
    for (int i = 0; i < 8; ++i)
        old_priv.push_back(cs.sub(old_v.inner[i], old_v.public_share[i]));
    StateWrapperVars new_v = old_v;
    std::vector<Var> new_priv = old_priv;
    new_v.inner[field] = cs.zero();
    std::vector<Var> pads, pubs_enc;
    stream_cipher_encrypt(cs, {new_v.inner[field]}, new_v.share, pads, pubs_enc);
    new_priv[field] = pads[0];
    new_v.public_share[field] = pubs_enc[0];
    cs.enforce_equal(pubs_enc[0], p_fee_share);
    Var rid = csprng_next(cs, new_v.recovery);
    cs.enforce_equal(rid, p_rid);
    Var old_comm = commitment_gadget(cs, old_priv, old_v.recovery, old_v.share,
                                     old_v.public_share);
    Var new_comm = commitment_gadget(cs, new_priv, new_v.recovery, new_v.share,
                                     new_v.public_share);
    cs.enforce_equal(new_comm, p_comm);
    Var root = merkle_root_gadget(cs, old_comm, op_elems, op_idx);
    cs.enforce_equal(root, p_root);
    Var nul = nullifier_gadget(cs, old_v);
    cs.enforce_equal(nul, p_null);
}

}  // namespace rng
