// PRODUCT PATH — MI355X-native BN254 field arithmetic (device + host).
//
// Replaces (new implementation, not a port): arkworks 0.4.2 ark-ff Montgomery
// arithmetic for BN254 Fr/Fq as consumed by the reference prover through
// mpc-jellyfish (limb layout pinned at
// crates/relayer-types/types-proofs/src/rkyv_impls/plonk_proof_def.rs:22-52).
// CIOS Montgomery multiplication with 4 x 64-bit limbs — deliberately a
// different formulation from the CPU oracle's SOS reduce so the two paths
// cross-check (oracle/field.hpp).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <cstring>
#include "../../include/bn254_params.h"

namespace rng {

using u64 = uint64_t;
using u128 = unsigned __int128;

#define RNG_HD __host__ __device__ __forceinline__

struct fr_params {
    static constexpr u64 mod[4] = FR_MODULUS;
    static constexpr u64 r[4] = FR_R;
    static constexpr u64 r2[4] = FR_R2;
    static constexpr u64 inv = FR_INV;
};
struct fq_params {
    static constexpr u64 mod[4] = FQ_MODULUS;
    static constexpr u64 r[4] = FQ_R;
    static constexpr u64 r2[4] = FQ_R2;
    static constexpr u64 inv = FQ_INV;
};

template <class P>
struct Fp4 {
    u64 l[4];

    RNG_HD static Fp4 zero() { return Fp4{{0, 0, 0, 0}}; }
    RNG_HD static Fp4 one() { return Fp4{{P::r[0], P::r[1], P::r[2], P::r[3]}}; }

    RNG_HD bool is_zero() const { return (l[0] | l[1] | l[2] | l[3]) == 0; }
    RNG_HD bool eq(const Fp4& o) const {
        return l[0] == o.l[0] && l[1] == o.l[1] && l[2] == o.l[2] && l[3] == o.l[3];
    }

    RNG_HD static bool geq_mod(const u64 a[4]) {
        for (int i = 3; i >= 0; --i) {
            if (a[i] != P::mod[i]) return a[i] > P::mod[i];
        }
        return true;
    }

    RNG_HD Fp4 add(const Fp4& o) const {
        Fp4 r;
        u128 c = 0;
        for (int i = 0; i < 4; ++i) {
            c += (u128)l[i] + o.l[i];
            r.l[i] = (u64)c;
            c >>= 64;
        }
        if (geq_mod(r.l)) {
            u128 b = 0;
            for (int i = 0; i < 4; ++i) {
                u128 d = (u128)r.l[i] - P::mod[i] - b;
                r.l[i] = (u64)d;
                b = (d >> 64) & 1;
            }
        }
        return r;
    }

    RNG_HD Fp4 sub(const Fp4& o) const {
        Fp4 r;
        u128 b = 0;
        for (int i = 0; i < 4; ++i) {
            u128 d = (u128)l[i] - o.l[i] - b;
            r.l[i] = (u64)d;
            b = (d >> 64) & 1;
        }
        if (b) {
            u128 c = 0;
            for (int i = 0; i < 4; ++i) {
                c += (u128)r.l[i] + P::mod[i];
                r.l[i] = (u64)c;
                c >>= 64;
            }
        }
        return r;
    }

    RNG_HD Fp4 neg() const {
        if (is_zero()) return *this;
        Fp4 r;
        u128 b = 0;
        for (int i = 0; i < 4; ++i) {
            u128 d = (u128)P::mod[i] - l[i] - b;
            r.l[i] = (u64)d;
            b = (d >> 64) & 1;
        }
        return r;
    }

    RNG_HD Fp4 dbl() const { return add(*this); }

    // Montgomery multiplication, N = 4: CIOS (Acar).  Formulation A/B on a
    // real MI355X (rng_bench_frmul, r02): CIOS-u64 78.6 Gmul/s, SOS-u64 57.7,
    // CIOS-u32 49.6 — despite SOS having ~1.5x fewer static issue slots, its
    // dynamically-indexed t[] spills into s_set_gpr_idx sequences and loses;
    // the u32 variant drowns in pack/unpack movs.  All variants are
    // bit-identical (pinned against each other and a Python bignum).
    RNG_HD Fp4 mul(const Fp4& b) const { return mul_cios(b); }

    // CIOS Montgomery multiplication (Acar), N = 4.
    RNG_HD Fp4 mul_cios(const Fp4& b) const {
        u64 t[6] = {0, 0, 0, 0, 0, 0};
        for (int i = 0; i < 4; ++i) {
            // t += a * b[i]
            u64 carry = 0;
            for (int j = 0; j < 4; ++j) {
                u128 cur = (u128)l[j] * b.l[i] + t[j] + carry;
                t[j] = (u64)cur;
                carry = (u64)(cur >> 64);
            }
            u128 cur = (u128)t[4] + carry;
            t[4] = (u64)cur;
            t[5] = (u64)(cur >> 64);
            // reduce one limb
            u64 m = t[0] * P::inv;
            cur = (u128)m * P::mod[0] + t[0];
            carry = (u64)(cur >> 64);
            for (int j = 1; j < 4; ++j) {
                cur = (u128)m * P::mod[j] + t[j] + carry;
                t[j - 1] = (u64)cur;
                carry = (u64)(cur >> 64);
            }
            cur = (u128)t[4] + carry;
            t[3] = (u64)cur;
            t[4] = t[5] + (u64)(cur >> 64);
        }
        Fp4 r{{t[0], t[1], t[2], t[3]}};
        if (t[4] || geq_mod(r.l)) {
            u128 bw = 0;
            for (int i = 0; i < 4; ++i) {
                u128 d = (u128)r.l[i] - P::mod[i] - bw;
                r.l[i] = (u64)d;
                bw = (d >> 64) & 1;
            }
        }
        return r;
    }

    // CIOS with 32-bit words: every inner step is a single
    // 32x32+32+32 -> 64 multiply-add (v_mad_u64_u32 shape) with the carry in
    // the same u64 — shorter carry chains than the 64-bit formulation.
    RNG_HD Fp4 mul32(const Fp4& o) const {
        uint32_t a[8], b[8], p[8], t[10];
        memcpy(a, l, 32);
        memcpy(b, o.l, 32);
        memcpy(p, P::mod, 32);
        for (int i = 0; i < 10; ++i) t[i] = 0;
        const uint32_t inv32 = (uint32_t)P::inv;
        for (int i = 0; i < 8; ++i) {
            uint64_t cur = 0;
            uint32_t carry = 0;
            for (int j = 0; j < 8; ++j) {
                cur = (uint64_t)a[j] * b[i] + t[j] + carry;
                t[j] = (uint32_t)cur;
                carry = (uint32_t)(cur >> 32);
            }
            cur = (uint64_t)t[8] + carry;
            t[8] = (uint32_t)cur;
            t[9] = t[9] + (uint32_t)(cur >> 32);
            uint32_t m = t[0] * inv32;
            cur = (uint64_t)m * p[0] + t[0];
            carry = (uint32_t)(cur >> 32);
            for (int j = 1; j < 8; ++j) {
                cur = (uint64_t)m * p[j] + t[j] + carry;
                t[j - 1] = (uint32_t)cur;
                carry = (uint32_t)(cur >> 32);
            }
            cur = (uint64_t)t[8] + carry;
            t[7] = (uint32_t)cur;
            t[8] = t[9] + (uint32_t)(cur >> 32);
            t[9] = 0;
        }
        Fp4 r;
        memcpy(r.l, t, 32);
        if (t[8] || geq_mod(r.l)) {
            u128 bw = 0;
            for (int i = 0; i < 4; ++i) {
                u128 d = (u128)r.l[i] - P::mod[i] - bw;
                r.l[i] = (u64)d;
                bw = (d >> 64) & 1;
            }
        }
        return r;
    }

    // 32-bit CIOS variant B: t[] and carry kept as u64 values with a
    // maintained hi=0 invariant, so each inner step is one v_mad_u64_u32
    // (32x32 + 64-bit addend) plus one 64-bit add — no zero-extension mov
    // dance.  (a*b + t + c <= (2^32-1)^2 + 2*(2^32-1) = 2^64-1: no overflow.)
    RNG_HD Fp4 mul_cios32b(const Fp4& o) const {
        uint32_t a[8], b[8], p[8];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            a[2 * i] = (uint32_t)l[i];
            a[2 * i + 1] = (uint32_t)(l[i] >> 32);
            b[2 * i] = (uint32_t)o.l[i];
            b[2 * i + 1] = (uint32_t)(o.l[i] >> 32);
            p[2 * i] = (uint32_t)P::mod[i];
            p[2 * i + 1] = (uint32_t)(P::mod[i] >> 32);
        }
        const uint32_t inv32 = (uint32_t)P::inv;
        u64 t[9];
#pragma unroll
        for (int i = 0; i < 9; ++i) t[i] = 0;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
            const u64 bi = b[i];
            // multiply pipeline: t += a * b[i]
            u64 cm = 0;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                u64 r = (u64)a[j] * bi + t[j] + cm;
                t[j] = (uint32_t)r;
                cm = r >> 32;
            }
            u64 top = t[8] + cm;
            // reduce pipeline: one limb of Montgomery reduction
            const u64 m = (uint32_t)((uint32_t)t[0] * inv32);
            u64 cr = ((u64)(uint32_t)m * p[0] + t[0]) >> 32;
#pragma unroll
            for (int j = 1; j < 8; ++j) {
                u64 r = (u64)(uint32_t)m * p[j] + t[j] + cr;
                t[j - 1] = (uint32_t)r;
                cr = r >> 32;
            }
            u64 r = top + cr;
            t[7] = (uint32_t)r;
            t[8] = r >> 32;
        }
        Fp4 res{{t[0] | (t[1] << 32), t[2] | (t[3] << 32), t[4] | (t[5] << 32),
                 t[6] | (t[7] << 32)}};
        if (t[8] || geq_mod(res.l)) {
            u128 bw = 0;
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                u128 d = (u128)res.l[i] - P::mod[i] - (u64)bw;
                res.l[i] = (u64)d;
                bw = (d >> 64) & 1;
            }
        }
        return res;
    }

    // 32-bit CIOS variant C: per row, all 8 multiply mads are issued into
    // INDEPENDENT 64-bit accumulators (no serial carry through the mads),
    // then one 32-bit carry-resolve chain folds the highs forward.
    RNG_HD Fp4 mul_cios32c(const Fp4& o) const {
        uint32_t a[8], b[8], p[8];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            a[2 * i] = (uint32_t)l[i];
            a[2 * i + 1] = (uint32_t)(l[i] >> 32);
            b[2 * i] = (uint32_t)o.l[i];
            b[2 * i + 1] = (uint32_t)(o.l[i] >> 32);
            p[2 * i] = (uint32_t)P::mod[i];
            p[2 * i + 1] = (uint32_t)(P::mod[i] >> 32);
        }
        const uint32_t inv32 = (uint32_t)P::inv;
        u64 t[9];
#pragma unroll
        for (int i = 0; i < 9; ++i) t[i] = 0;
        u64 r[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
            const u64 bi = b[i];
#pragma unroll
            for (int j = 0; j < 8; ++j) r[j] = (u64)a[j] * bi + t[j];  // independent
            // resolve: r[j] < 2^64 - 2^32; adding a 32-bit carry cannot overflow
            u64 c = 0;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                r[j] += c;
                t[j] = (uint32_t)r[j];
                c = r[j] >> 32;
            }
            u64 top = t[8] + c;
            const u64 m = (uint32_t)((uint32_t)t[0] * inv32);
#pragma unroll
            for (int j = 0; j < 8; ++j) r[j] = (u64)(uint32_t)m * p[j] + t[j];
            c = r[0] >> 32;
#pragma unroll
            for (int j = 1; j < 8; ++j) {
                r[j] += c;
                t[j - 1] = (uint32_t)r[j];
                c = r[j] >> 32;
            }
            u64 rr = top + c;
            t[7] = (uint32_t)rr;
            t[8] = rr >> 32;
        }
        Fp4 res{{t[0] | (t[1] << 32), t[2] | (t[3] << 32), t[4] | (t[5] << 32),
                 t[6] | (t[7] << 32)}};
        if (t[8] || geq_mod(res.l)) {
            u128 bw = 0;
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                u128 d = (u128)res.l[i] - P::mod[i] - (u64)bw;
                res.l[i] = (u64)d;
                bw = (d >> 64) & 1;
            }
        }
        return res;
    }

    // SOS: full 4x4 product (independent partial products -> ILP), then a
    // separate 4-round Montgomery reduction.
    RNG_HD Fp4 mul_sos(const Fp4& b) const {
        u64 t[8];
        u64 carry = 0;
        // i = 0 row initializes
        {
            u128 cur;
            cur = (u128)l[0] * b.l[0];
            t[0] = (u64)cur;
            carry = (u64)(cur >> 64);
            cur = (u128)l[1] * b.l[0] + carry;
            t[1] = (u64)cur;
            carry = (u64)(cur >> 64);
            cur = (u128)l[2] * b.l[0] + carry;
            t[2] = (u64)cur;
            carry = (u64)(cur >> 64);
            cur = (u128)l[3] * b.l[0] + carry;
            t[3] = (u64)cur;
            t[4] = (u64)(cur >> 64);
        }
        for (int i = 1; i < 4; ++i) {
            u64 c2 = 0;
            for (int j = 0; j < 4; ++j) {
                u128 cur = (u128)l[j] * b.l[i] + t[i + j] + c2;
                t[i + j] = (u64)cur;
                c2 = (u64)(cur >> 64);
            }
            t[i + 4] = c2;
        }
        // Montgomery reduce
        u64 extra = 0;
        for (int i = 0; i < 4; ++i) {
            u64 m = t[i] * P::inv;
            u64 c2 = 0;
            for (int j = 0; j < 4; ++j) {
                u128 cur = (u128)m * P::mod[j] + t[i + j] + c2;
                t[i + j] = (u64)cur;
                c2 = (u64)(cur >> 64);
            }
            // propagate into t[i+4]
            u128 cur = (u128)t[i + 4] + c2 + ((i > 0) ? 0 : 0);
            // also absorb previous extra at the top limb
            cur += (i + 4 == 7) ? 0 : 0;
            t[i + 4] = (u64)cur;
            u64 c3 = (u64)(cur >> 64);
            for (int k = i + 5; k < 8 && c3; ++k) {
                u128 c4 = (u128)t[k] + c3;
                t[k] = (u64)c4;
                c3 = (u64)(c4 >> 64);
            }
            extra += (i + 5 > 7) ? c3 : 0;
        }
        Fp4 r{{t[4], t[5], t[6], t[7]}};
        if (extra || geq_mod(r.l)) {
            u128 bw = 0;
            for (int i = 0; i < 4; ++i) {
                u128 d = (u128)r.l[i] - P::mod[i] - bw;
                r.l[i] = (u64)d;
                bw = (d >> 64) & 1;
            }
        }
        return r;
    }

    RNG_HD Fp4 sqr() const { return mul(*this); }

    RNG_HD Fp4 pow(const u64 e[4]) const {
        Fp4 acc = one();
        Fp4 base = *this;
        for (int i = 0; i < 256; ++i) {
            if ((e[i >> 6] >> (i & 63)) & 1) acc = acc.mul(base);
            base = base.sqr();
        }
        return acc;
    }
    RNG_HD Fp4 pow_u64(u64 e) const {
        Fp4 acc = one();
        Fp4 base = *this;
        while (e) {
            if (e & 1) acc = acc.mul(base);
            base = base.sqr();
            e >>= 1;
        }
        return acc;
    }
    RNG_HD Fp4 inverse() const {  // Fermat: a^(p-2); low limb of both moduli >= 2
        u64 e[4] = {P::mod[0] - 2, P::mod[1], P::mod[2], P::mod[3]};
        return pow(e);
    }

    RNG_HD static Fp4 from_u64(u64 v) {
        u64 c[4] = {v, 0, 0, 0};
        return from_canonical(c);
    }
    RNG_HD static Fp4 from_canonical(const u64 c[4]) {
        Fp4 a{{c[0], c[1], c[2], c[3]}};
        Fp4 r2{{P::r2[0], P::r2[1], P::r2[2], P::r2[3]}};
        return a.mul(r2);
    }
    RNG_HD void to_canonical(u64 out[4]) const {
        // multiply by 1 (non-Montgomery) = Montgomery-reduce
        Fp4 onev{{1, 0, 0, 0}};
        Fp4 r = mul(onev);
        out[0] = r.l[0]; out[1] = r.l[1]; out[2] = r.l[2]; out[3] = r.l[3];
    }
};

using Fr = Fp4<fr_params>;
using Fq = Fp4<fq_params>;

}  // namespace rng
