// PRODUCT PATH — BN254 pairing for proof verification (host).
//
// Replaces the pairing check inside mpc-plonk's PlonkKzgSnark::verify
// (called at circuit-types/src/traits.rs:1012) and backs the SRS
// well-formedness check mirroring srs.rs:236-266.
//
// Implementation: reduced TATE pairing e(P, Q) = f_{r,P}(psi(Q))^((q^12-1)/r)
// with the Miller loop over the G1 argument (381 line steps over Fq) and the
// final exponentiation as one big power (the verifier is a CPU self-check
// path, not the proving hot loop — ~tens of ms per verification is fine and
// avoids hand-derived Frobenius constants entirely).  The tower is
// Fq2 = Fq[u]/(u^2+1), Fq6 = Fq2[v]/(v^3 - xi), xi = 9+u,
// Fq12 = Fq6[w]/(w^2 - v); the sextic twist embeds G2 via
// psi(x', y') = (x' v, y' v w)  (E': y'^2 = x'^3 + 3/xi, D-twist).
#pragma once
#include <vector>
#include "gpu_field.hpp"

namespace rng {

struct PFq2 {
    Fq a, b;  // a + b u
    static PFq2 zero() { return {Fq::zero(), Fq::zero()}; }
    static PFq2 one() { return {Fq::one(), Fq::zero()}; }
    bool is_zero() const { return a.is_zero() && b.is_zero(); }
    bool eq(const PFq2& o) const { return a.eq(o.a) && b.eq(o.b); }
    PFq2 add(const PFq2& o) const { return {a.add(o.a), b.add(o.b)}; }
    PFq2 sub(const PFq2& o) const { return {a.sub(o.a), b.sub(o.b)}; }
    PFq2 neg() const { return {a.neg(), b.neg()}; }
    PFq2 mul(const PFq2& o) const {
        Fq v0 = a.mul(o.a), v1 = b.mul(o.b);
        return {v0.sub(v1), a.add(b).mul(o.a.add(o.b)).sub(v0).sub(v1)};
    }
    PFq2 sqr() const {
        Fq t = a.mul(b);
        return {a.add(b).mul(a.sub(b)), t.dbl()};
    }
    PFq2 scale(const Fq& s) const { return {a.mul(s), b.mul(s)}; }
    PFq2 inverse() const {
        Fq norm = a.sqr().add(b.sqr());
        Fq ni = norm.inverse();
        return {a.mul(ni), b.neg().mul(ni)};
    }
    static PFq2 xi() {  // 9 + u
        return {Fq::from_u64(9), Fq::one()};
    }
    PFq2 mul_xi() const {  // (a+bu)(9+u) = 9a - b + (a + 9b)u
        Fq nine_a = a.mul(Fq::from_u64(9));
        Fq nine_b = b.mul(Fq::from_u64(9));
        return {nine_a.sub(b), a.add(nine_b)};
    }
};

struct PFq6 {  // c0 + c1 v + c2 v^2
    PFq2 c0, c1, c2;
    static PFq6 zero() { return {PFq2::zero(), PFq2::zero(), PFq2::zero()}; }
    static PFq6 one() { return {PFq2::one(), PFq2::zero(), PFq2::zero()}; }
    bool is_zero() const { return c0.is_zero() && c1.is_zero() && c2.is_zero(); }
    PFq6 add(const PFq6& o) const { return {c0.add(o.c0), c1.add(o.c1), c2.add(o.c2)}; }
    PFq6 sub(const PFq6& o) const { return {c0.sub(o.c0), c1.sub(o.c1), c2.sub(o.c2)}; }
    PFq6 neg() const { return {c0.neg(), c1.neg(), c2.neg()}; }
    PFq6 mul(const PFq6& o) const {
        PFq2 v0 = c0.mul(o.c0), v1 = c1.mul(o.c1), v2 = c2.mul(o.c2);
        PFq2 t0 = c1.add(c2).mul(o.c1.add(o.c2)).sub(v1).sub(v2).mul_xi().add(v0);
        PFq2 t1 = c0.add(c1).mul(o.c0.add(o.c1)).sub(v0).sub(v1).add(v2.mul_xi());
        PFq2 t2 = c0.add(c2).mul(o.c0.add(o.c2)).sub(v0).sub(v2).add(v1);
        return {t0, t1, t2};
    }
    PFq6 sqr() const { return mul(*this); }
    PFq6 mul_v() const {  // * v : (c0,c1,c2) -> (xi*c2, c0, c1)
        return {c2.mul_xi(), c0, c1};
    }
    PFq6 inverse() const {
        // standard: A = c0^2 - xi c1 c2, B = xi c2^2 - c0 c1, C = c1^2 - c0 c2
        PFq2 A = c0.sqr().sub(c1.mul(c2).mul_xi());
        PFq2 B = c2.sqr().mul_xi().sub(c0.mul(c1));
        PFq2 C = c1.sqr().sub(c0.mul(c2));
        PFq2 F = c2.mul(B).add(c1.mul(C)).mul_xi().add(c0.mul(A));
        PFq2 Fi = F.inverse();
        return {A.mul(Fi), B.mul(Fi), C.mul(Fi)};
    }
};

struct PFq12 {  // d0 + d1 w
    PFq6 d0, d1;
    static PFq12 one() { return {PFq6::one(), PFq6::zero()}; }
    bool eq(const PFq12& o) const {
        auto eq6 = [](const PFq6& x, const PFq6& y) {
            return x.c0.eq(y.c0) && x.c1.eq(y.c1) && x.c2.eq(y.c2);
        };
        return eq6(d0, o.d0) && eq6(d1, o.d1);
    }
    PFq12 mul(const PFq12& o) const {
        PFq6 v0 = d0.mul(o.d0), v1 = d1.mul(o.d1);
        PFq6 t1 = d0.add(d1).mul(o.d0.add(o.d1)).sub(v0).sub(v1);
        return {v0.add(v1.mul_v()), t1};
    }
    PFq12 sqr() const { return mul(*this); }
    PFq12 inverse() const {
        PFq6 t = d0.sqr().sub(d1.sqr().mul_v());
        PFq6 ti = t.inverse();
        return {d0.mul(ti), d1.neg().mul(ti)};
    }
};

// G2 affine in Fq2 (twist curve coordinates as stored in the SRS:
// x.c0, x.c1, y.c0, y.c1 Montgomery limbs)
struct PG2 {
    PFq2 x, y;
};

// Miller loop f_{r,P}(psi(Q)) with the loop over r on G1.
// Line through A=(ax,ay), B (both G1, affine) evaluated at
// psi(Q) = (qx v, qy v w):
//   chord/tangent y = lam x + c: value = qy v w - lam qx v - c
//   -> Fq12 coords: d0.c1 = -lam*qx (Fq2), d0.c0 = -c, d1.c1 = qy
struct MillerCtx {
    PFq2 qx, qy;
    PFq12 line(const Fq& lam, const Fq& c) const {
        PFq12 r{PFq6::zero(), PFq6::zero()};
        r.d0.c0 = PFq2{c.neg(), Fq::zero()};
        r.d0.c1 = qx.scale(lam).neg();
        r.d1.c1 = qy;
        return r;
    }
    // vertical line x = a: value = qx v - a
    PFq12 vertical(const Fq& a) const {
        PFq12 r{PFq6::zero(), PFq6::zero()};
        r.d0.c0 = PFq2{a.neg(), Fq::zero()};
        r.d0.c1 = qx;
        return r;
    }
};

// Reduced Tate pairing.  P = (px, py) affine G1 (not infinity), Q affine G2.
PFq12 tate_pairing(const Fq& px, const Fq& py, const PG2& q);

// multi-check helper: e(p1, q1) == e(p2, q2)
bool pairing_check_eq(const Fq& p1x, const Fq& p1y, const PG2& q1, const Fq& p2x,
                      const Fq& p2y, const PG2& q2);

}  // namespace rng
