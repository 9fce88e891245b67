// PRODUCT PATH — Tate pairing implementation (see pairing.hpp header).
#pragma once
#include "pairing.hpp"
#include "../../include/bn254_final_exp.h"

namespace rng {

inline PFq12 fq12_pow_limbs(const PFq12& base, const uint64_t* e, int nlimbs) {
    PFq12 acc = PFq12::one();
    PFq12 b = base;
    for (int limb = 0; limb < nlimbs; ++limb) {
        for (int bit = 0; bit < 64; ++bit) {
            if ((e[limb] >> bit) & 1) acc = acc.mul(b);
            b = b.sqr();
        }
    }
    return acc;
}

inline PFq12 final_exponentiation(const PFq12& f) {
    return fq12_pow_limbs(f, BN254_FINAL_EXP, BN254_FINAL_EXP_LIMBS);
}

// Miller loop over r (MSB-first double-and-add on G1), denominators (vertical
// lines) eliminated: with psi(Q) = (qx v, qy v w) a vertical's value lies in
// Fq6*, which the final exponent (q^6-1)(q^6+1)/r kills.
inline PFq12 miller_tate(const Fq& px, const Fq& py, const MillerCtx& mc) {
    static const u64 rmod[4] = FR_MODULUS;
    // bits of r, MSB first
    int top = 253;  // r < 2^254; bit 253 is the MSB
    Fq tx = px, ty = py;
    PFq12 f = PFq12::one();
    bool t_inf = false;
    for (int i = top - 1; i >= 0; --i) {
        // doubling: tangent at T
        if (!t_inf) {
            Fq lam = tx.sqr().mul(Fq::from_u64(3)).mul(ty.dbl().inverse());
            Fq c = ty.sub(lam.mul(tx));
            f = f.sqr().mul(mc.line(lam, c));
            // T = 2T
            Fq x3 = lam.sqr().sub(tx.dbl());
            Fq y3 = lam.mul(tx.sub(x3)).sub(ty);
            tx = x3;
            ty = y3;
        } else {
            f = f.sqr();
        }
        if ((rmod[i >> 6] >> (i & 63)) & 1) {
            if (t_inf) continue;
            if (tx.eq(px)) {
                if (ty.eq(py)) {
                    // T == P: tangent case (cannot happen mid-loop for prime r)
                    Fq lam = tx.sqr().mul(Fq::from_u64(3)).mul(ty.dbl().inverse());
                    Fq c = ty.sub(lam.mul(tx));
                    f = f.mul(mc.line(lam, c));
                    Fq x3 = lam.sqr().sub(tx.dbl());
                    Fq y3 = lam.mul(tx.sub(x3)).sub(ty);
                    tx = x3;
                    ty = y3;
                } else {
                    // T == -P: chord is the vertical line (eliminated); T -> O
                    t_inf = true;
                }
            } else {
                Fq lam = ty.sub(py).mul(tx.sub(px).inverse());
                Fq c = py.sub(lam.mul(px));
                f = f.mul(mc.line(lam, c));
                Fq x3 = lam.sqr().sub(tx).sub(px);
                Fq y3 = lam.mul(tx.sub(x3)).sub(ty);
                tx = x3;
                ty = y3;
            }
        }
    }
    return f;
}

PFq12 tate_pairing(const Fq& px, const Fq& py, const PG2& q) {
    MillerCtx mc{q.x, q.y};
    return final_exponentiation(miller_tate(px, py, mc));
}

bool pairing_check_eq(const Fq& p1x, const Fq& p1y, const PG2& q1, const Fq& p2x,
                      const Fq& p2y, const PG2& q2) {
    // e(P1, Q1) == e(P2, Q2) <=> miller(P1,Q1) * miller(P2,-Q2) ^ finalexp == 1
    MillerCtx m1{q1.x, q1.y};
    MillerCtx m2{q2.x, q2.y.neg()};
    PFq12 f = miller_tate(p1x, p1y, m1).mul(miller_tate(p2x, p2y, m2));
    return final_exponentiation(f).eq(PFq12::one());
}

}  // namespace rng
