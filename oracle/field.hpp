// CPU ORACLE — test infrastructure only.
//
// This file is part of the CPU restatement ("oracle") of the reference's
// PlonK-prover arithmetic.  Only tests/, __graft_entry__.smoke() and
// bench.py's cpu_baseline leg may link or execute this code; the product
// path (renegade_amd + librenegade_prover.so) never routes through it.
//
// Restates: arkworks 0.4.2 ark-ff Montgomery backend semantics for BN254
// Fr/Fq as pinned in-repo by the reference at
//   crates/relayer-types/types-proofs/src/rkyv_impls/plonk_proof_def.rs:22-52
//   (4 x u64 little-endian Montgomery limbs)
//   crates/crypto/src/fields.rs:21-26 (moduli)
// arkworks itself is a non-vendored crates.io dep (Cargo.lock:933-1199) and
// cannot be built in this container (no Rust toolchain): at the proof-byte
// level parity is therefore pinned against this oracle, which is itself
// validated against an independent Python-bignum model in tests/
// (see SURVEY.md §8c — "parity unpinned" against the reference binary).
#pragma once
#include <cstdint>
#include <cstring>
#include <array>
#include "../include/bn254_params.h"

namespace oracle {

using u64 = uint64_t;
using u128 = unsigned __int128;

struct FpParams {
    u64 mod[4];
    u64 r[4];
    u64 r2[4];
    u64 inv;
};

inline constexpr FpParams FQ_P = {FQ_MODULUS, FQ_R, FQ_R2, FQ_INV};
inline constexpr FpParams FR_P = {FR_MODULUS, FR_R, FR_R2, FR_INV};

// 4x64 helpers (plain, carry-explicit; SOS-style Montgomery reduction --
// deliberately a different formulation from the HIP path's CIOS so the two
// implementations cross-check rather than share bugs).
template <const FpParams& P>
struct Fp {
    u64 l[4];  // little-endian limbs, Montgomery form (value = l * 2^-256 mod p)

    static inline bool geq(const u64 a[4], const u64 b[4]) {
        for (int i = 3; i >= 0; --i) {
            if (a[i] != b[i]) return a[i] > b[i];
        }
        return true;
    }
    static inline void sub4(u64 out[4], const u64 a[4], const u64 b[4]) {
        u128 borrow = 0;
        for (int i = 0; i < 4; ++i) {
            u128 d = (u128)a[i] - b[i] - borrow;
            out[i] = (u64)d;
            borrow = (d >> 64) & 1;  // two's-complement borrow propagates
        }
    }
    static inline u64 add4(u64 out[4], const u64 a[4], const u64 b[4]) {
        u128 carry = 0;
        for (int i = 0; i < 4; ++i) {
            u128 s = (u128)a[i] + b[i] + carry;
            out[i] = (u64)s;
            carry = s >> 64;
        }
        return (u64)carry;
    }

    static Fp zero() { Fp f; f.l[0] = f.l[1] = f.l[2] = f.l[3] = 0; return f; }
    static Fp one() { Fp f; memcpy(f.l, P.r, 32); return f; }

    bool is_zero() const { return (l[0] | l[1] | l[2] | l[3]) == 0; }
    bool operator==(const Fp& o) const { return memcmp(l, o.l, 32) == 0; }
    bool operator!=(const Fp& o) const { return !(*this == o); }

    Fp operator+(const Fp& o) const {
        Fp r;
        u64 c = add4(r.l, l, o.l);
        if (c || geq(r.l, P.mod)) {
            u64 t[4];
            sub4(t, r.l, P.mod);
            memcpy(r.l, t, 32);
        }
        return r;
    }
    Fp operator-(const Fp& o) const {
        Fp r;
        if (geq(l, o.l)) {
            sub4(r.l, l, o.l);
        } else {
            u64 t[4];
            add4(t, l, P.mod);
            sub4(r.l, t, o.l);
        }
        return r;
    }
    Fp neg() const {
        if (is_zero()) return *this;
        Fp r;
        sub4(r.l, P.mod, l);
        return r;
    }
    Fp dbl() const { return *this + *this; }

    // Montgomery multiplication: full 512-bit product then Montgomery reduce.
    Fp operator*(const Fp& o) const {
        u64 t[8] = {0};
        for (int i = 0; i < 4; ++i) {
            u64 carry = 0;
            for (int j = 0; j < 4; ++j) {
                u128 cur = (u128)l[i] * o.l[j] + t[i + j] + carry;
                t[i + j] = (u64)cur;
                carry = (u64)(cur >> 64);
            }
            t[i + 4] = carry;
        }
        return reduce(t);
    }
    Fp square() const { return *this * *this; }

    // Montgomery reduction of a 512-bit value t (< p * 2^256).
    static Fp reduce(const u64 tin[8]) {
        u64 t[9];
        memcpy(t, tin, 64);
        t[8] = 0;
        for (int i = 0; i < 4; ++i) {
            u64 m = t[i] * P.inv;
            u64 carry = 0;
            for (int j = 0; j < 4; ++j) {
                u128 cur = (u128)m * P.mod[j] + t[i + j] + carry;
                t[i + j] = (u64)cur;
                carry = (u64)(cur >> 64);
            }
            // propagate carry into upper limbs (t[8] absorbs the top)
            for (int k = i + 4; k < 9 && carry; ++k) {
                u128 cur = (u128)t[k] + carry;
                t[k] = (u64)cur;
                carry = (u64)(cur >> 64);
            }
        }
        Fp r;
        memcpy(r.l, t + 4, 32);
        if (t[8] || geq(r.l, P.mod)) {
            u64 s[4];
            sub4(s, r.l, P.mod);
            memcpy(r.l, s, 32);
        }
        return r;
    }

    // Conversion: canonical integer (LE limbs) <-> Montgomery form
    static Fp from_canonical(const u64 c[4]) {
        Fp a;
        memcpy(a.l, c, 32);
        Fp r2;
        memcpy(r2.l, P.r2, 32);
        return a * r2;
    }
    void to_canonical(u64 out[4]) const {
        u64 t[8] = {0};
        memcpy(t, l, 32);
        Fp r = reduce(t);
        memcpy(out, r.l, 32);
    }
    static Fp from_u64(u64 v) {
        u64 c[4] = {v, 0, 0, 0};
        return from_canonical(c);
    }

    Fp pow(const u64 e[4]) const {  // LSB-first square-and-multiply
        Fp acc = one();
        Fp base = *this;
        for (int i = 0; i < 256; ++i) {
            int limb = i / 64, bit = i % 64;
            if ((e[limb] >> bit) & 1) acc = acc * base;
            base = base.square();
        }
        return acc;
    }
    Fp pow_u64(u64 e) const {
        u64 ee[4] = {e, 0, 0, 0};
        return pow(ee);
    }
    Fp inverse() const {  // Fermat: a^(p-2)
        u64 e[4];
        memcpy(e, P.mod, 32);
        // e = p - 2
        u64 two[4] = {2, 0, 0, 0};
        sub4(e, e, two);
        return pow(e);
    }
};

using Fq = Fp<FQ_P>;
using Fr = Fp<FR_P>;

}  // namespace oracle
