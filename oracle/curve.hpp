// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Restates: arkworks 0.4.2 ark-ec short-Weierstrass BN254 G1 group math, as
// consumed by the reference's KZG commitments.
// Affine representation pinned in-repo at
//   crates/relayer-types/types-proofs/src/rkyv_impls/plonk_proof_def.rs:70-100
//   (G1Affine { x: Fq, y: Fq, infinity: bool }, Montgomery limbs).
// Formulas are the standard Jacobian add/double (EFD); the group is what is
// pinned, not the internal coordinate system.
#pragma once
#include "field.hpp"

namespace oracle {

struct G1Affine {
    Fq x, y;
    bool infinity;
    static G1Affine identity() { return {Fq::zero(), Fq::zero(), true}; }
    static G1Affine generator() {
        G1Affine g;
        static constexpr u64 gx[4] = G1_GEN_X_MONT, gy[4] = G1_GEN_Y_MONT;
        memcpy(g.x.l, gx, 32);
        memcpy(g.y.l, gy, 32);
        g.infinity = false;
        return g;
    }
    bool is_on_curve() const {
        if (infinity) return true;
        Fq b;
        static constexpr u64 bm[4] = G1_B_MONT;
        memcpy(b.l, bm, 32);
        return y.square() == x.square() * x + b;
    }
    bool operator==(const G1Affine& o) const {
        if (infinity || o.infinity) return infinity == o.infinity;
        return x == o.x && y == o.y;
    }
};

// Jacobian: (X, Y, Z) with x = X/Z^2, y = Y/Z^3; identity encoded Z = 0.
struct G1Proj {
    Fq X, Y, Z;
    static G1Proj identity() { return {Fq::one(), Fq::one(), Fq::zero()}; }
    bool is_identity() const { return Z.is_zero(); }

    static G1Proj from_affine(const G1Affine& a) {
        if (a.infinity) return identity();
        return {a.x, a.y, Fq::one()};
    }
    G1Affine to_affine() const {
        if (is_identity()) return G1Affine::identity();
        Fq zinv = Z.inverse();
        Fq zinv2 = zinv.square();
        return {X * zinv2, Y * zinv2 * zinv, false};
    }

    G1Proj dbl() const {
        if (is_identity()) return *this;
        // EFD dbl-2007-bl: A=X1^2, B=Y1^2, C=B^2, D=2((X1+B)^2-A-C), E=3A, F=E^2
        Fq A = X.square();
        Fq B = Y.square();
        Fq C = B.square();
        Fq D = ((X + B).square() - A - C).dbl();
        Fq E = A + A + A;
        Fq F = E.square();
        G1Proj r;
        r.X = F - D.dbl();
        r.Y = E * (D - r.X) - C.dbl().dbl().dbl();
        r.Z = (Y * Z).dbl();
        return r;
    }

    G1Proj add(const G1Proj& o) const {
        if (is_identity()) return o;
        if (o.is_identity()) return *this;
        // EFD add-2007-bl
        Fq Z1Z1 = Z.square();
        Fq Z2Z2 = o.Z.square();
        Fq U1 = X * Z2Z2;
        Fq U2 = o.X * Z1Z1;
        Fq S1 = Y * o.Z * Z2Z2;
        Fq S2 = o.Y * Z * Z1Z1;
        if (U1 == U2) {
            if (S1 == S2) return dbl();
            return identity();
        }
        Fq H = U2 - U1;
        Fq I = H.dbl().square();
        Fq J = H * I;
        Fq rr = (S2 - S1).dbl();
        Fq V = U1 * I;
        G1Proj r;
        r.X = rr.square() - J - V.dbl();
        r.Y = rr * (V - r.X) - (S1 * J).dbl();
        r.Z = ((Z + o.Z).square() - Z1Z1 - Z2Z2) * H;
        return r;
    }

    // Mixed addition: other point affine (Z2 = 1). EFD madd-2007-bl.
    G1Proj add_affine(const G1Affine& o) const {
        if (o.infinity) return *this;
        if (is_identity()) return from_affine(o);
        Fq Z1Z1 = Z.square();
        Fq U2 = o.x * Z1Z1;
        Fq S2 = o.y * Z * Z1Z1;
        if (U2 == X) {
            if (S2 == Y) return dbl();
            return identity();
        }
        Fq H = U2 - X;
        Fq HH = H.square();
        Fq I = HH.dbl().dbl();
        Fq J = H * I;
        Fq rr = (S2 - Y).dbl();
        Fq V = X * I;
        G1Proj r;
        r.X = rr.square() - J - V.dbl();
        r.Y = rr * (V - r.X) - (Y * J).dbl();
        r.Z = (Z + H).square() - Z1Z1 - HH;
        return r;
    }

    G1Proj neg() const { return {X, Y.neg(), Z}; }

    // Scalar multiplication by canonical (non-Montgomery) little-endian limbs.
    G1Proj mul(const u64 e[4]) const {
        G1Proj acc = identity();
        for (int i = 255; i >= 0; --i) {
            acc = acc.dbl();
            if ((e[i / 64] >> (i % 64)) & 1) acc = acc.add(*this);
        }
        return acc;
    }
};

}  // namespace oracle
