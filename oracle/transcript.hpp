// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Fiat-Shamir transcript + deterministic blinder DRBG.
//
// Restates the ROLE of mpc-plonk's keccak256-based `SolidityTranscript`
// (consumed at crates/circuits/circuit-types/src/traits.rs:996,1012).  The
// reference's exact byte layout lives in the non-vendored mpc-jellyfish
// sources (Cargo.lock:6064) and no in-repo test pins transcript bytes
// (parity at proof-byte level vs the reference binary is
// UNPINNED in this container).  The byte layout below is therefore the
// normative spec FOR THIS BUILD, implemented identically (and independently)
// by the product prover; parity = bit-exact proofs between the two on the
// same SRS/witness/seed, + verifier acceptance.
//
// Spec (shared with renegade_amd/csrc/plonk_host.hpp):
//   state: 32 bytes, initially all zero.  buf: bytes appended since the last
//   challenge.  append_u64 -> 8 B LE; append_fr -> 32 B LE canonical;
//   append_g1 -> x||y, each 32 B LE canonical (identity -> 64 zero bytes).
//   challenge(): h = keccak256(state || buf); state = h; buf.clear();
//   return LE-integer(h) mod r.
//
//   Blinder DRBG: block(i) = keccak256("rng-blind" || le64(seed) || le32(i));
//   b_i = LE-integer(block(i)) mod r, i = 0,1,2,...
#pragma once
#include <vector>
#include "field.hpp"
#include "curve.hpp"
#include "keccak.hpp"

namespace oracle {

inline Fr fr_from_hash_le(const uint8_t h[32]) {
    // LE 256-bit integer mod r: from_canonical performs the full reduction
    // (multiplication by R2 reduces any 256-bit input)
    u64 limbs[4];
    memcpy(limbs, h, 32);
    return Fr::from_canonical(limbs);
}

struct Transcript {
    uint8_t state[32];
    std::vector<uint8_t> buf;

    Transcript() { memset(state, 0, 32); }

    void append_bytes(const uint8_t* p, size_t n) { buf.insert(buf.end(), p, p + n); }
    void append_u64(u64 x) { append_bytes((uint8_t*)&x, 8); }
    void append_fr(const Fr& f) {
        u64 c[4];
        f.to_canonical(c);
        append_bytes((uint8_t*)c, 32);
    }
    void append_fq(const Fq& f) {
        u64 c[4];
        f.to_canonical(c);
        append_bytes((uint8_t*)c, 32);
    }
    void append_g1(const G1Affine& p) {
        if (p.infinity) {
            uint8_t z[64] = {0};
            append_bytes(z, 64);
        } else {
            append_fq(p.x);
            append_fq(p.y);
        }
    }
    Fr challenge() {
        Keccak256 k;
        k.update(state, 32);
        if (!buf.empty()) k.update(buf.data(), buf.size());
        uint8_t h[32];
        k.finalize(h);
        memcpy(state, h, 32);
        buf.clear();
        return fr_from_hash_le(h);
    }
};

struct BlinderDrbg {
    u64 seed;
    uint32_t ctr = 0;
    explicit BlinderDrbg(u64 s) : seed(s) {}
    Fr next() {
        uint8_t msg[9 + 8 + 4];
        memcpy(msg, "rng-blind", 9);
        memcpy(msg + 9, &seed, 8);
        memcpy(msg + 17, &ctr, 4);
        ctr++;
        uint8_t h[32];
        keccak256(msg, 21, h);
        return fr_from_hash_le(h);
    }
};

}  // namespace oracle
