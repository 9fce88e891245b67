// PRODUCT PATH — deterministic synthetic circuits for parity tests/bench.
//
// Exercises the full arithmetization API surface (the census in SURVEY.md
// §8a) the way the reference's gadgets do: arithmetic gates, booleans,
// muxes, range checks (boolean bits + lc_sum reconstruction, like
// zk_gadgets/primitives/bits.rs), fused power-5 gates (like the Poseidon2
// gates), and public inputs.
#pragma once
#include "plonk_circuit.hpp"

namespace rng {

struct Lcg {
    uint64_t s;
    explicit Lcg(uint64_t seed) : s(seed * 6364136223846793005ull + 1442695040888963407ull) {}
    uint64_t next() {
        s = s * 6364136223846793005ull + 1442695040888963407ull;
        return s >> 11;
    }
    Fr fr() {
        u64 l[4] = {next() | (next() << 53), next() | (next() << 53),
                    next() | (next() << 53), next() & ((1ull << 60) - 1)};
        return Fr::from_canonical(l);  // < 2^252 < r
    }
};

// Builds a mixed circuit with ~(24 * scale) gates and 4 public inputs.
inline void build_mixed_circuit(PlonkCircuit& cs, uint64_t seed, uint64_t scale) {
    Lcg rng(seed);
    std::vector<Var> pool;
    for (int i = 0; i < 8; ++i) pool.push_back(cs.create_variable(rng.fr()));

    for (uint64_t it = 0; it < scale; ++it) {
        Var a = pool[rng.next() % pool.size()];
        Var b = pool[rng.next() % pool.size()];
        pool.push_back(cs.add(a, b));
        pool.push_back(cs.mul(a, b));
        pool.push_back(cs.sub(pool.back(), a));
        pool.push_back(cs.add_constant(b, rng.fr()));
        pool.push_back(cs.mul_constant(a, rng.fr()));
        pool.push_back(cs.lc({a, b, pool[0], pool[1]},
                             {rng.fr(), rng.fr(), rng.fr(), rng.fr()}));
        pool.push_back(cs.mul_add({a, b, pool[2], pool[3]}, {rng.fr(), rng.fr()}));
        // booleans + logic
        Var bit0 = cs.create_boolean_variable((rng.next() & 1) ? Fr::one() : Fr::zero());
        Var bit1 = cs.create_boolean_variable((rng.next() & 1) ? Fr::one() : Fr::zero());
        pool.push_back(cs.mux(bit0, a, b));
        Var andv = cs.logic_and(bit0, bit1);
        Var orv = cs.logic_or(bit0, bit1);
        Var negv = cs.logic_neg(andv);
        cs.enforce_bool(orv);
        cs.enforce_bool(negv);
        // range check on a fresh small value (like AmountGadget)
        Var small = cs.create_variable(Fr::from_u64(rng.next() & 0xFFFFFF));
        cs.enforce_in_range(small, 24);
        // fused pow-5 gate (Poseidon2-style, q_hash selectors)
        {
            Var x0 = a, x1 = b, x2 = pool[4 % pool.size()], x3 = pool[5 % pool.size()];
            auto p5 = [&](Var x) {
                Fr v = cs.witness(x);
                Fr v2 = v.sqr();
                return v2.sqr().mul(v);
            };
            Fr rc = rng.fr();
            Fr out = rc.add(p5(x0)).add(p5(x1)).add(p5(x2)).add(p5(x3));
            Var o = cs.create_variable(out);
            std::array<Fr, NUM_SELECTORS> q{};
            for (auto& s : q) s = Fr::zero();
            q[SEL_HASH0] = q[SEL_HASH1] = q[SEL_HASH2] = q[SEL_HASH3] = Fr::one();
            q[SEL_C] = rc;
            q[SEL_O] = Fr::one();
            cs.insert_gate({x0, x1, x2, x3, o}, q);
            pool.push_back(o);
        }
        // copy constraint stress: reuse a value in a later lc
        cs.enforce_equal(pool.back(), pool.back());
        if (pool.size() > 64) pool.erase(pool.begin(), pool.begin() + 32);
    }
    // public inputs: expose four derived values
    for (int i = 0; i < 4 && i < (int)pool.size(); ++i) cs.set_public(pool[pool.size() - 1 - i]);
}

}  // namespace rng
