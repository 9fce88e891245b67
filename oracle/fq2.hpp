// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Fq2 = Fq[u]/(u^2 + 1) and the BN254 G2 group, as required by the SRS's
// (h, beta_h) pair (reference parser:
//  crates/circuits/circuit-types/src/primitives/srs.rs:147-198) and by the
// CPU pairing used for KZG verification.
#pragma once
#include "field.hpp"

namespace oracle {

struct Fq2 {
    Fq c0, c1;  // c0 + c1*u
    static Fq2 zero() { return {Fq::zero(), Fq::zero()}; }
    static Fq2 one() { return {Fq::one(), Fq::zero()}; }
    bool is_zero() const { return c0.is_zero() && c1.is_zero(); }
    bool operator==(const Fq2& o) const { return c0 == o.c0 && c1 == o.c1; }
    bool operator!=(const Fq2& o) const { return !(*this == o); }
    Fq2 operator+(const Fq2& o) const { return {c0 + o.c0, c1 + o.c1}; }
    Fq2 operator-(const Fq2& o) const { return {c0 - o.c0, c1 - o.c1}; }
    Fq2 neg() const { return {c0.neg(), c1.neg()}; }
    Fq2 dbl() const { return {c0.dbl(), c1.dbl()}; }
    Fq2 operator*(const Fq2& o) const {
        // Karatsuba: (a0+a1 u)(b0+b1 u) = a0b0 - a1b1 + ((a0+a1)(b0+b1)-a0b0-a1b1) u
        Fq v0 = c0 * o.c0;
        Fq v1 = c1 * o.c1;
        Fq2 r;
        r.c0 = v0 - v1;
        r.c1 = (c0 + c1) * (o.c0 + o.c1) - v0 - v1;
        return r;
    }
    Fq2 square() const {
        // (a0 + a1 u)^2 = (a0+a1)(a0-a1) + 2 a0 a1 u
        Fq2 r;
        Fq t = c0 * c1;
        r.c0 = (c0 + c1) * (c0 - c1);
        r.c1 = t.dbl();
        return r;
    }
    Fq2 scale(const Fq& s) const { return {c0 * s, c1 * s}; }
    Fq2 conjugate() const { return {c0, c1.neg()}; }
    Fq2 inverse() const {
        // 1/(a0 + a1 u) = (a0 - a1 u) / (a0^2 + a1^2)
        Fq norm = c0.square() + c1.square();
        Fq ninv = norm.inverse();
        return {c0 * ninv, c1.neg() * ninv};
    }
    Fq2 mul_by_u() const { return {c1.neg(), c0}; }  // * u, u^2 = -1
};

struct G2Affine {
    Fq2 x, y;
    bool infinity;
    static G2Affine identity() { return {Fq2::zero(), Fq2::zero(), true}; }
    static G2Affine generator() {
        static constexpr u64 x0[4] = G2_GEN_X_C0_MONT, x1[4] = G2_GEN_X_C1_MONT,
                             y0[4] = G2_GEN_Y_C0_MONT, y1[4] = G2_GEN_Y_C1_MONT;
        G2Affine g;
        memcpy(g.x.c0.l, x0, 32); memcpy(g.x.c1.l, x1, 32);
        memcpy(g.y.c0.l, y0, 32); memcpy(g.y.c1.l, y1, 32);
        g.infinity = false;
        return g;
    }
    bool is_on_curve() const {
        if (infinity) return true;
        static constexpr u64 b0[4] = G2_B_C0_MONT, b1[4] = G2_B_C1_MONT;
        Fq2 b;
        memcpy(b.c0.l, b0, 32); memcpy(b.c1.l, b1, 32);
        return y.square() == x.square() * x + b;
    }
    bool operator==(const G2Affine& o) const {
        if (infinity || o.infinity) return infinity == o.infinity;
        return x == o.x && y == o.y;
    }
};

struct G2Proj {  // Jacobian
    Fq2 X, Y, Z;
    static G2Proj identity() { return {Fq2::one(), Fq2::one(), Fq2::zero()}; }
    bool is_identity() const { return Z.is_zero(); }
    static G2Proj from_affine(const G2Affine& a) {
        if (a.infinity) return identity();
        return {a.x, a.y, Fq2::one()};
    }
    G2Affine to_affine() const {
        if (is_identity()) return G2Affine::identity();
        Fq2 zinv = Z.inverse();
        Fq2 zinv2 = zinv.square();
        return {X * zinv2, Y * zinv2 * zinv, false};
    }
    G2Proj dbl() const {
        if (is_identity()) return *this;
        Fq2 A = X.square(), B = Y.square(), C = B.square();
        Fq2 D = ((X + B).square() - A - C).dbl();
        Fq2 E = A + A + A, F = E.square();
        G2Proj r;
        r.X = F - D.dbl();
        r.Y = E * (D - r.X) - C.dbl().dbl().dbl();
        r.Z = (Y * Z).dbl();
        return r;
    }
    G2Proj add(const G2Proj& o) const {
        if (is_identity()) return o;
        if (o.is_identity()) return *this;
        Fq2 Z1Z1 = Z.square(), Z2Z2 = o.Z.square();
        Fq2 U1 = X * Z2Z2, U2 = o.X * Z1Z1;
        Fq2 S1 = Y * o.Z * Z2Z2, S2 = o.Y * Z * Z1Z1;
        if (U1 == U2) {
            if (S1 == S2) return dbl();
            return identity();
        }
        Fq2 H = U2 - U1, I = H.dbl().square(), J = H * I;
        Fq2 rr = (S2 - S1).dbl(), V = U1 * I;
        G2Proj r;
        r.X = rr.square() - J - V.dbl();
        r.Y = rr * (V - r.X) - (S1 * J).dbl();
        r.Z = ((Z + o.Z).square() - Z1Z1 - Z2Z2) * H;
        return r;
    }
    G2Proj mul(const u64 e[4]) const {  // canonical scalar
        G2Proj acc = identity();
        for (int i = 255; i >= 0; --i) {
            acc = acc.dbl();
            if ((e[i / 64] >> (i % 64)) & 1) acc = acc.add(*this);
        }
        return acc;
    }
};

}  // namespace oracle
