// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Independent Poseidon2 restatement (crypto/src/hash/poseidon2.rs:89-208;
// constants pinned at crypto/src/hash/constants.rs ->
// include/poseidon2_constants.h).  Cross-checks the product's native
// implementation in tests; the reference's own known-answer test compares
// against the zkhash crate at runtime (a non-vendored git dep —
// poseidon2.rs:211-284), so the constants + structure are the pin here.
#pragma once
#include "field.hpp"
#include "../include/poseidon2_constants.h"

namespace oracle {

struct Poseidon2Sponge {
    Fr st[3];
    int idx = 0;
    bool squeezing = false;

    Poseidon2Sponge() { st[0] = st[1] = st[2] = Fr::zero(); }

    static Fr pow5(const Fr& x) {
        Fr x2 = x.square();
        return x2.square() * x;
    }
    void ext_mds() {
        Fr s = st[0] + st[1] + st[2];
        st[0] = st[0] + s;
        st[1] = st[1] + s;
        st[2] = st[2] + s;
    }
    void int_mds() {
        Fr s = st[0] + st[1] + st[2];
        st[2] = st[2].dbl();
        st[0] = st[0] + s;
        st[1] = st[1] + s;
        st[2] = st[2] + s;
    }
    void permute() {
        ext_mds();
        for (int r = 0; r < POSEIDON2_R_F / 2; ++r) {
            for (int w = 0; w < 3; ++w)
                st[w] = st[w] + Fr::from_canonical(POSEIDON2_FULL_RC[r][w]);
            for (int w = 0; w < 3; ++w) st[w] = pow5(st[w]);
            ext_mds();
        }
        for (int r = 0; r < POSEIDON2_R_P; ++r) {
            st[0] = st[0] + Fr::from_canonical(POSEIDON2_PARTIAL_RC[r]);
            st[0] = pow5(st[0]);
            int_mds();
        }
        for (int r = POSEIDON2_R_F / 2; r < POSEIDON2_R_F; ++r) {
            for (int w = 0; w < 3; ++w)
                st[w] = st[w] + Fr::from_canonical(POSEIDON2_FULL_RC[r][w]);
            for (int w = 0; w < 3; ++w) st[w] = pow5(st[w]);
            ext_mds();
        }
    }
    void absorb(const Fr& x) {
        if (idx == 2) {
            permute();
            idx = 0;
        }
        st[idx + 1] = st[idx + 1] + x;
        idx++;
    }
    Fr squeeze() {
        if (!squeezing || idx == 2) {
            permute();
            idx = 0;
            squeezing = true;
        }
        return st[1 + idx++];
    }
};

inline Fr poseidon2_hash(const Fr* in, size_t n) {
    Poseidon2Sponge s;
    for (size_t i = 0; i < n; ++i) s.absorb(in[i]);
    return s.squeeze();
}

}  // namespace oracle
