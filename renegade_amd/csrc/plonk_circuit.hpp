// PRODUCT PATH — TurboPlonk constraint system (host).
//
// New implementation of the arithmetization layer the reference consumes from
// mpc-relation (non-vendored dep, mpc-jellyfish@311568a; surface census in
// SURVEY.md §8a "Arithmetization API"):  5 wires (GATE_WIDTH=4 inputs + 1
// output), 13 selectors, copy constraints via an extended permutation over
// 5n wire slots, public-input gates first, power-of-two padding
// (finalize_for_arithmetization), proof-linking groups.
//
// Gate equation (pinned in-repo by the reference's own Gate impls at
// crates/circuits/circuits-core/src/zk_gadgets/primitives/poseidon/gates.rs:
// 73-101,150-179 — q_lc/q_mul/q_hash/q_c/q_o semantics):
//   q_c + PI(X) + sum_i q_lc[i] w_i + q_mul[0] w0 w1 + q_mul[1] w2 w3
//       + sum_i q_hash[i] w_i^5 + q_ecc w0 w1 w2 w3 w4 - q_o w4  =  0
#pragma once
#include <algorithm>
#include <array>
#include <cassert>
#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

#include "gpu_field.hpp"

namespace rng {

constexpr int NUM_WIRES = 5;
constexpr int NUM_SELECTORS = 13;
// selector column order (also the C-ABI layout, include/rng_prover.h):
enum Sel {
    SEL_LC0 = 0, SEL_LC1, SEL_LC2, SEL_LC3,
    SEL_MUL0, SEL_MUL1,
    SEL_HASH0, SEL_HASH1, SEL_HASH2, SEL_HASH3,
    SEL_O, SEL_C, SEL_ECC,
};

using Var = uint32_t;

struct GateRow {
    std::array<Var, NUM_WIRES> wires{};       // variable ids
    std::array<Fr, NUM_SELECTORS> q{};        // selector values (Montgomery)
};

// Proof-linking group (mpc-relation GroupLayout semantics): the group's
// wire-0 values are placed on the shared subgroup H_{2^alignment} at grid
// positions [offset, offset+size), i.e. gate indices (offset+k) * (n/2^a).
// Because w_n^(n/2^a) = w_{2^a} for every domain n, two circuits of
// DIFFERENT sizes that place a group at the same (alignment, offset) put its
// values at the same field points — which is what cross-circuit linking
// opens against.
struct LinkGroup {
    std::string id;
    int alignment = -1;   // grid = 2^alignment points; -1 = auto-assign
    int64_t offset = -1;  // grid offset; -1 = auto-assign
    std::vector<Var> vars;
};

// Finalized tables (feeds RngCircuitDesc / the provers)
struct CircuitTables {
    uint64_t n = 0;             // padded domain size (power of two)
    uint64_t num_public = 0;
    std::vector<Fr> selectors;  // 13 columns * n, column-major [sel*n + gate]
    std::vector<uint64_t> sigma;  // 5*n slot permutation: sigma[j*n+i] = j'*n+i'
    std::vector<Fr> wires;      // 5 columns * n wire VALUES [j*n + i]
    std::vector<Fr> public_inputs;
    // link groups: id -> (alignment, grid offset, count); gate index of
    // element k = (offset+k) * (n >> alignment)
    struct PlacedGroup {
        std::string id;
        uint64_t alignment, offset, count;
    };
    std::vector<PlacedGroup> link_groups;
};

class PlonkCircuit {
  public:
    PlonkCircuit() {
        // var 0 = constant 0, var 1 = constant 1 (mirrors mpc-relation's
        // zero()/one() wiring)
        values_.push_back(Fr::zero());
        values_.push_back(Fr::one());
        enforce_constant(0, Fr::zero());
        enforce_constant(1, Fr::one());
    }

    // ---- variable management ----
    Var create_variable(const Fr& v) {
        values_.push_back(v);
        return (Var)(values_.size() - 1);
    }
    Var create_public_variable(const Fr& v) {
        Var x = create_variable(v);
        set_public(x);
        return x;
    }
    void set_public(Var x) { public_vars_.push_back(x); }
    Var create_boolean_variable(const Fr& v) {
        Var x = create_variable(v);
        enforce_bool(x);
        return x;
    }
    Var zero() const { return 0; }
    Var one() const { return 1; }
    const Fr& witness(Var x) const { return values_.at(x); }
    size_t num_vars() const { return values_.size(); }
    size_t num_gates() const { return gates_.size(); }

    // ---- raw gate insertion (Gate trait analogue) ----
    void insert_gate(const std::array<Var, NUM_WIRES>& wires,
                     const std::array<Fr, NUM_SELECTORS>& q) {
        for (Var w : wires) assert(w < values_.size());
        gates_.push_back(GateRow{wires, q});
    }

    // ---- constraints ----
    void enforce_equal(Var a, Var b) {
        auto q = zq();
        q[SEL_LC0] = Fr::one();
        q[SEL_LC1] = Fr::one().neg();
        insert_gate({a, b, 0, 0, 0}, q);
    }
    void enforce_constant(Var a, const Fr& c) {
        auto q = zq();
        q[SEL_LC0] = Fr::one();
        q[SEL_C] = c.neg();
        insert_gate({a, 0, 0, 0, 0}, q);
    }
    void enforce_bool(Var a) {  // a^2 = a
        auto q = zq();
        q[SEL_MUL0] = Fr::one();
        q[SEL_O] = Fr::one();
        insert_gate({a, a, 0, 0, a}, q);
    }
    void enforce_true(Var a) { enforce_constant(a, Fr::one()); }
    void enforce_false(Var a) { enforce_constant(a, Fr::zero()); }

    // ---- arithmetic gates (each returns the output variable) ----
    Var add(Var a, Var b) {
        Fr v = witness(a).add(witness(b));
        Var o = create_variable(v);
        auto q = zq();
        q[SEL_LC0] = q[SEL_LC1] = Fr::one();
        q[SEL_O] = Fr::one();
        insert_gate({a, b, 0, 0, o}, q);
        return o;
    }
    Var sub(Var a, Var b) {
        Fr v = witness(a).sub(witness(b));
        Var o = create_variable(v);
        auto q = zq();
        q[SEL_LC0] = Fr::one();
        q[SEL_LC1] = Fr::one().neg();
        q[SEL_O] = Fr::one();
        insert_gate({a, b, 0, 0, o}, q);
        return o;
    }
    Var mul(Var a, Var b) {
        Fr v = witness(a).mul(witness(b));
        Var o = create_variable(v);
        auto q = zq();
        q[SEL_MUL0] = Fr::one();
        q[SEL_O] = Fr::one();
        insert_gate({a, b, 0, 0, o}, q);
        return o;
    }
    Var add_constant(Var a, const Fr& c) {
        Fr v = witness(a).add(c);
        Var o = create_variable(v);
        auto q = zq();
        q[SEL_LC0] = Fr::one();
        q[SEL_C] = c;
        q[SEL_O] = Fr::one();
        insert_gate({a, 0, 0, 0, o}, q);
        return o;
    }
    Var mul_constant(Var a, const Fr& c) {
        Fr v = witness(a).mul(c);
        Var o = create_variable(v);
        auto q = zq();
        q[SEL_LC0] = c;
        q[SEL_O] = Fr::one();
        insert_gate({a, 0, 0, 0, o}, q);
        return o;
    }
    // out = sum coeffs[i] * vars[i] (4-ary linear combination)
    Var lc(const std::array<Var, 4>& vars, const std::array<Fr, 4>& coeffs) {
        Fr v = Fr::zero();
        for (int i = 0; i < 4; ++i) v = v.add(witness(vars[i]).mul(coeffs[i]));
        Var o = create_variable(v);
        auto q = zq();
        for (int i = 0; i < 4; ++i) q[SEL_LC0 + i] = coeffs[i];
        q[SEL_O] = Fr::one();
        insert_gate({vars[0], vars[1], vars[2], vars[3], o}, q);
        return o;
    }
    // sum of arbitrarily many vars: chain of 4-ary lcs (bits.rs:30-100 style)
    Var lc_sum(const std::vector<Var>& vars) {
        if (vars.empty()) return 0;
        std::array<Fr, 4> ones{Fr::one(), Fr::one(), Fr::one(), Fr::one()};
        Var acc = 0;
        size_t i = 0;
        // first group of up to 4
        std::array<Var, 4> grp{0, 0, 0, 0};
        for (int k = 0; k < 4 && i < vars.size(); ++k, ++i) grp[k] = vars[i];
        acc = lc(grp, ones);
        while (i < vars.size()) {
            std::array<Var, 4> g2{acc, 0, 0, 0};
            for (int k = 1; k < 4 && i < vars.size(); ++k, ++i) g2[k] = vars[i];
            acc = lc(g2, ones);
        }
        return acc;
    }
    // out = m0*a*b + m1*c*d
    Var mul_add(const std::array<Var, 4>& v, const std::array<Fr, 2>& m) {
        Fr val = m[0].mul(witness(v[0]).mul(witness(v[1])))
                     .add(m[1].mul(witness(v[2]).mul(witness(v[3]))));
        Var o = create_variable(val);
        auto q = zq();
        q[SEL_MUL0] = m[0];
        q[SEL_MUL1] = m[1];
        q[SEL_O] = Fr::one();
        insert_gate({v[0], v[1], v[2], v[3], o}, q);
        return o;
    }
    // out = b ? x : y  (b boolean)  = b*x - b*y + y
    Var mux(Var b, Var x, Var y) {
        Fr vb = witness(b);
        Fr val = vb.mul(witness(x)).add(Fr::one().sub(vb).mul(witness(y)));
        Var o = create_variable(val);
        // gate: b*x - b*y + y - o = 0 -> q_mul0(w0,w1)=1, q_mul1(w2,w3)=-1,
        // q_lc3(w3=y)=1
        auto q = zq();
        q[SEL_MUL0] = Fr::one();
        q[SEL_MUL1] = Fr::one().neg();
        q[SEL_LC3] = Fr::one();
        q[SEL_O] = Fr::one();
        insert_gate({b, x, b, y, o}, q);
        return o;
    }
    Var logic_neg(Var b) {  // 1 - b
        Fr val = Fr::one().sub(witness(b));
        Var o = create_variable(val);
        auto q = zq();
        q[SEL_LC0] = Fr::one().neg();
        q[SEL_C] = Fr::one();
        q[SEL_O] = Fr::one();
        insert_gate({b, 0, 0, 0, o}, q);
        return o;
    }
    Var logic_and(Var a, Var b) { return mul(a, b); }
    Var logic_and_all(const std::vector<Var>& vs) {
        // product of booleans == 1 iff all true; chain muls
        if (vs.empty()) return 1;
        Var acc = vs[0];
        for (size_t i = 1; i < vs.size(); ++i) acc = mul(acc, vs[i]);
        return acc;
    }
    Var logic_or(Var a, Var b) {  // a + b - ab
        Fr va = witness(a), vb = witness(b);
        Fr val = va.add(vb).sub(va.mul(vb));
        Var o = create_variable(val);
        auto q = zq();
        q[SEL_LC0] = Fr::one();
        q[SEL_LC1] = Fr::one();
        q[SEL_MUL0] = Fr::one().neg();
        q[SEL_O] = Fr::one();
        insert_gate({a, b, 0, 0, o}, q);
        return o;
    }

    // ---- bit decomposition / range checks (bits.rs / bitlength.rs) ----
    // allocate `bits` boolean vars for x (little-endian) and constrain
    // reconstruction; x must fit in `bits` bits for satisfiability.
    std::vector<Var> to_bits(Var x, int bits) {
        u64 limbs[4];
        witness(x).to_canonical(limbs);
        std::vector<Var> bv;
        bv.reserve(bits);
        for (int i = 0; i < bits; ++i) {
            u64 bit = (limbs[i / 64] >> (i % 64)) & 1;
            bv.push_back(create_boolean_variable(bit ? Fr::one() : Fr::zero()));
        }
        // reconstruct with powers of two via 4-ary lcs
        Fr two = Fr::from_u64(2);
        std::vector<Var> terms;
        Var acc = 0;
        Fr p = Fr::one();
        size_t i = 0;
        bool first = true;
        while (i < bv.size()) {
            std::array<Var, 4> g{acc, 0, 0, 0};
            std::array<Fr, 4> cf{Fr::one(), Fr::zero(), Fr::zero(), Fr::zero()};
            int start = first ? 0 : 1;
            for (int k = start; k < 4 && i < bv.size(); ++k, ++i) {
                g[k] = bv[i];
                cf[k] = p;
                p = p.mul(two);
            }
            acc = lc(g, cf);
            first = false;
        }
        enforce_equal(acc, x);
        return bv;
    }
    void enforce_in_range(Var x, int bits) { (void)to_bits(x, bits); }

    // ---- proof linking ----
    void create_link_group(const std::string& id, int alignment = -1,
                           int64_t offset = -1) {
        link_groups_.push_back(LinkGroup{id, alignment, offset, {}});
    }
    void add_to_link_group(Var v, const std::string& id) {
        for (auto& g : link_groups_) {
            if (g.id == id) {
                g.vars.push_back(v);
                return;
            }
        }
        throw std::runtime_error("unknown link group " + id);
    }

    // ---- satisfiability check (mirrors check_circuit_satisfiability) ----
    bool check_satisfied(std::string* why = nullptr) const {
        // public-input gates are implicit: pub var handling happens via the
        // PI polynomial; here check every explicit gate with PI = 0 and the
        // public gates as (w0 - pub) = 0 which is trivially true since the
        // value IS the variable's value.
        for (size_t i = 0; i < gates_.size(); ++i) {
            if (!gate_satisfied(gates_[i], Fr::zero())) {
                if (why) *why = "gate " + std::to_string(i) + " unsatisfied";
                return false;
            }
        }
        return true;
    }

    // ---- finalize (finalize_for_arithmetization) ----
    // Layout: [public-input gates][explicit gates][link-group gates][padding]
    CircuitTables finalize();

  private:
    static std::array<Fr, NUM_SELECTORS> zq() {
        std::array<Fr, NUM_SELECTORS> q;
        for (auto& x : q) x = Fr::zero();
        return q;
    }
    bool gate_satisfied(const GateRow& g, const Fr& pi) const {
        const Fr w0 = values_[g.wires[0]], w1 = values_[g.wires[1]],
                 w2 = values_[g.wires[2]], w3 = values_[g.wires[3]],
                 w4 = values_[g.wires[4]];
        Fr r = g.q[SEL_C].add(pi);
        r = r.add(g.q[SEL_LC0].mul(w0)).add(g.q[SEL_LC1].mul(w1));
        r = r.add(g.q[SEL_LC2].mul(w2)).add(g.q[SEL_LC3].mul(w3));
        r = r.add(g.q[SEL_MUL0].mul(w0.mul(w1))).add(g.q[SEL_MUL1].mul(w2.mul(w3)));
        auto p5 = [](const Fr& x) {
            Fr x2 = x.sqr();
            return x2.sqr().mul(x);
        };
        r = r.add(g.q[SEL_HASH0].mul(p5(w0))).add(g.q[SEL_HASH1].mul(p5(w1)));
        r = r.add(g.q[SEL_HASH2].mul(p5(w2))).add(g.q[SEL_HASH3].mul(p5(w3)));
        r = r.add(g.q[SEL_ECC].mul(w0.mul(w1).mul(w2).mul(w3).mul(w4)));
        r = r.sub(g.q[SEL_O].mul(w4));
        return r.is_zero();
    }

    std::vector<Fr> values_;
    std::vector<GateRow> gates_;
    std::vector<Var> public_vars_;
    std::vector<LinkGroup> link_groups_;
};

inline CircuitTables PlonkCircuit::finalize() {
    CircuitTables T;
    size_t n_pub = public_vars_.size();

    // main gate order: pub gates first, then the explicit gates
    std::vector<GateRow> main;
    main.reserve(n_pub + gates_.size());
    for (Var pv : public_vars_) {
        auto q = zq();
        q[SEL_LC0] = Fr::one().neg();  // -pub + PI(w^i) = 0
        main.push_back(GateRow{{pv, 0, 0, 0, 0}, q});
    }
    for (auto& g : gates_) main.push_back(g);

    // ---- link-group placement on the alignment grid ----
    uint64_t total_link = 0;
    for (auto& lg : link_groups_) total_link += lg.vars.size();
    // auto alignment: smallest grid holding offset-1 start + all groups
    int auto_align = 0;
    while ((1ull << auto_align) < total_link + 2) auto_align++;
    int align = auto_align;
    for (auto& lg : link_groups_)
        if (lg.alignment > align) align = lg.alignment;
    // assign grid offsets (consecutive from 1) where unspecified
    {
        int64_t next_off = 1;
        for (auto& lg : link_groups_) {
            if (lg.offset >= 0) {
                next_off = std::max(next_off, lg.offset + (int64_t)lg.vars.size());
            }
        }
        int64_t cursor = 1;
        for (auto& lg : link_groups_) {
            if (lg.offset < 0) {
                lg.offset = cursor;
                cursor += (int64_t)lg.vars.size();
            } else {
                cursor = std::max(cursor, lg.offset + (int64_t)lg.vars.size());
            }
        }
        if ((1ll << align) < cursor) {
            while ((1ll << align) < cursor) align++;
        }
    }

    // domain size: must fit main gates + link gates and give stride > n_pub
    uint64_t n = 8;
    auto fits = [&](uint64_t nn) {
        if (nn < main.size() + total_link) return false;
        if (link_groups_.empty()) return true;
        if (nn < (1ull << align)) return false;
        uint64_t stride = nn >> align;
        if (stride <= n_pub) return false;  // grid slots must avoid PI gates
        return true;
    };
    while (!fits(n)) n <<= 1;

    // lay out: reserved slots for link gates, others fill in order
    std::vector<GateRow> all(n);
    GateRow pad{};
    pad.wires = {0, 0, 0, 0, 0};
    pad.q = zq();
    for (auto& g : all) g = pad;
    std::vector<bool> reserved(n, false);
    std::vector<CircuitTables::PlacedGroup> placed;
    uint64_t stride = link_groups_.empty() ? 0 : (n >> align);
    for (auto& lg : link_groups_) {
        for (size_t k = 0; k < lg.vars.size(); ++k) {
            uint64_t idx = (uint64_t)(lg.offset + (int64_t)k) * stride;
            auto q = zq();  // all-zero selectors: trivially satisfied
            all[idx] = GateRow{{lg.vars[k], 0, 0, 0, 0}, q};
            reserved[idx] = true;
        }
        placed.push_back({lg.id, (uint64_t)align, (uint64_t)lg.offset, lg.vars.size()});
    }
    {
        uint64_t slot = 0;
        for (auto& g : main) {
            while (reserved[slot]) ++slot;
            all[slot++] = g;
        }
        // note: PI gates land at 0..n_pub-1 because stride > n_pub keeps the
        // reserved grid slots out of that prefix
    }

    T.n = n;
    T.num_public = n_pub;
    T.selectors.assign(NUM_SELECTORS * n, Fr::zero());
    T.wires.assign(NUM_WIRES * n, Fr::zero());
    T.sigma.assign(NUM_WIRES * n, 0);
    T.link_groups = std::move(placed);
    for (uint64_t i = 0; i < n; ++i) {
        for (int s = 0; s < NUM_SELECTORS; ++s) T.selectors[s * n + i] = all[i].q[s];
        for (int j = 0; j < NUM_WIRES; ++j) T.wires[j * n + i] = values_[all[i].wires[j]];
    }
    T.public_inputs.reserve(n_pub);
    for (Var pv : public_vars_) T.public_inputs.push_back(values_[pv]);

    // permutation: slots with the same variable form one cycle.
    // slot id = j*n + i; sigma[slot] = next slot in the variable's cycle.
    {
        std::vector<int64_t> first(values_.size(), -1), prev(values_.size(), -1);
        for (int j = 0; j < NUM_WIRES; ++j) {
            for (uint64_t i = 0; i < n; ++i) {
                Var v = all[i].wires[j];
                int64_t slot = (int64_t)j * n + i;
                if (first[v] < 0) {
                    first[v] = slot;
                } else {
                    T.sigma[prev[v]] = slot;
                }
                prev[v] = slot;
            }
        }
        for (size_t v = 0; v < values_.size(); ++v) {
            if (first[v] >= 0) T.sigma[prev[v]] = first[v];  // close the cycle
        }
    }
    return T;
}

// coset representatives k_j for the 5-wire permutation argument: k_0 = 1 and
// k_j = delta^j with delta chosen (deterministically, shared by prover and
// verifier) so the five cosets k_j * H are disjoint for every domain size we
// support (n <= 2^26): delta=7 is verified at plan time by assert_cosets().
inline void coset_ks(Fr out[NUM_WIRES]) {
    out[0] = Fr::one();
    Fr d = Fr::from_u64(7);
    Fr acc = Fr::one();
    for (int j = 1; j < NUM_WIRES; ++j) {
        acc = acc.mul(d);
        out[j] = acc;
    }
}

inline bool cosets_ok(uint64_t n) {
    // k_i H distinct <=> (k_i/k_j)^n != 1 for i != j <=> 7^(d*n) != 1, d=1..4
    Fr d7 = Fr::from_u64(7);
    for (uint64_t d = 1; d <= 4; ++d) {
        Fr x = d7.pow_u64(d * n);
        // pow over full exponent: d*n <= 2^28 fits u64
        if (x.eq(Fr::one())) return false;
    }
    return true;
}

}  // namespace rng
