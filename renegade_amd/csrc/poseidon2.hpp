// PRODUCT PATH — Poseidon2 sponge over BN254 Fr (native, host).
//
// Restates crates/crypto/src/hash/poseidon2.rs:89-208 (t=3, R_F=8, R_P=56,
// rate 2, capacity 1, alpha=5; external MDS circ(2,1,1), internal MDS
// [[2,1,1],[1,2,1],[1,1,3]]); round constants pinned in-repo at
// crypto/src/hash/constants.rs (include/poseidon2_constants.h).
// Used for witness-side commitments/nullifiers — this is
// front-end (witness construction) work, not the GPU hot path.
#pragma once
#include "gpu_field.hpp"
#include "../../include/poseidon2_constants.h"

namespace rng {

struct Poseidon2 {
    static constexpr int WIDTH = 3, RATE = 2, CAPACITY = 1, R_F = 8, R_P = 56;
    Fr state[3];
    int next_index = 0;
    bool squeezing = false;

    Poseidon2() { state[0] = state[1] = state[2] = Fr::zero(); }

    static Fr rc_full(int r, int w) {
        return Fr::from_canonical(POSEIDON2_FULL_RC[r][w]);
    }
    static Fr rc_partial(int r) { return Fr::from_canonical(POSEIDON2_PARTIAL_RC[r]); }

    static Fr sbox(const Fr& x) {
        Fr x2 = x.sqr();
        return x2.sqr().mul(x);
    }
    void external_mds() {
        Fr sum = state[0].add(state[1]).add(state[2]);
        for (auto& s : state) s = s.add(sum);
    }
    void internal_mds() {
        Fr sum = state[0].add(state[1]).add(state[2]);
        state[2] = state[2].dbl();
        for (auto& s : state) s = s.add(sum);
    }
    void permute() {
        external_mds();
        for (int r = 0; r < R_F / 2; ++r) {
            for (int w = 0; w < 3; ++w) state[w] = state[w].add(rc_full(r, w));
            for (auto& s : state) s = sbox(s);
            external_mds();
        }
        for (int r = 0; r < R_P; ++r) {
            state[0] = state[0].add(rc_partial(r));
            state[0] = sbox(state[0]);
            internal_mds();
        }
        for (int r = R_F / 2; r < R_F; ++r) {
            for (int w = 0; w < 3; ++w) state[w] = state[w].add(rc_full(r, w));
            for (auto& s : state) s = sbox(s);
            external_mds();
        }
    }
    void absorb(const Fr& x) {
        // (poseidon2.rs:45-57)
        if (next_index == RATE) {
            permute();
            next_index = 0;
        }
        state[next_index + CAPACITY] = state[next_index + CAPACITY].add(x);
        next_index++;
    }
    Fr squeeze() {
        if (!squeezing || next_index == RATE) {
            permute();
            next_index = 0;
            squeezing = true;
        }
        return state[CAPACITY + next_index++];
    }
};

// compute_poseidon_hash (crypto/src/hash/mod.rs): absorb all, squeeze one
inline Fr poseidon_hash(const Fr* inputs, size_t n) {
    Poseidon2 sp;
    for (size_t i = 0; i < n; ++i) sp.absorb(inputs[i]);
    return sp.squeeze();
}

}  // namespace rng
