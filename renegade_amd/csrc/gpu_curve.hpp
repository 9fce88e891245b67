// PRODUCT PATH — MI355X-native BN254 G1 group ops (device + host).
//
// Replaces arkworks ark-ec G1 arithmetic on the KZG-commit hot path
// (SURVEY.md §8a a5).  Jacobian coordinates, standard EFD add/dbl/madd
// formulas; identity encoded as Z == 0.
#pragma once
#include "gpu_field.hpp"

namespace rng {

struct G1Aff {  // 8 u64: x, y Montgomery limbs. Points at infinity are not
                // representable here; the MSM contract excludes them.
    Fq x, y;
};

struct G1Jac {
    Fq X, Y, Z;

    RNG_HD static G1Jac identity() {
        G1Jac p;
        p.X = Fq::one();
        p.Y = Fq::one();
        p.Z = Fq::zero();
        return p;
    }
    RNG_HD bool is_identity() const { return Z.is_zero(); }

    RNG_HD static G1Jac from_affine(const G1Aff& a) {
        G1Jac p;
        p.X = a.x;
        p.Y = a.y;
        p.Z = Fq::one();
        return p;
    }

    RNG_HD G1Jac dbl() const {
        if (is_identity()) return *this;
        Fq A = X.sqr();
        Fq B = Y.sqr();
        Fq C = B.sqr();
        Fq D = X.add(B).sqr().sub(A).sub(C).dbl();
        Fq E = A.add(A).add(A);
        Fq F = E.sqr();
        G1Jac r;
        r.X = F.sub(D.dbl());
        r.Y = E.mul(D.sub(r.X)).sub(C.dbl().dbl().dbl());
        r.Z = Y.mul(Z).dbl();
        return r;
    }

    RNG_HD G1Jac add(const G1Jac& o) const {
        if (is_identity()) return o;
        if (o.is_identity()) return *this;
        Fq Z1Z1 = Z.sqr();
        Fq Z2Z2 = o.Z.sqr();
        Fq U1 = X.mul(Z2Z2);
        Fq U2 = o.X.mul(Z1Z1);
        Fq S1 = Y.mul(o.Z).mul(Z2Z2);
        Fq S2 = o.Y.mul(Z).mul(Z1Z1);
        if (U1.eq(U2)) {
            if (S1.eq(S2)) return dbl();
            return identity();
        }
        Fq H = U2.sub(U1);
        Fq I = H.dbl().sqr();
        Fq J = H.mul(I);
        Fq rr = S2.sub(S1).dbl();
        Fq V = U1.mul(I);
        G1Jac r;
        r.X = rr.sqr().sub(J).sub(V.dbl());
        r.Y = rr.mul(V.sub(r.X)).sub(S1.mul(J).dbl());
        r.Z = Z.add(o.Z).sqr().sub(Z1Z1).sub(Z2Z2).mul(H);
        return r;
    }

    // Mixed add (o affine, never infinity). `negate` adds -o.
    RNG_HD G1Jac madd(const G1Aff& o, bool negate = false) const {
        Fq oy = negate ? o.y.neg() : o.y;
        if (is_identity()) {
            G1Jac r;
            r.X = o.x;
            r.Y = oy;
            r.Z = Fq::one();
            return r;
        }
        Fq Z1Z1 = Z.sqr();
        Fq U2 = o.x.mul(Z1Z1);
        Fq S2 = oy.mul(Z).mul(Z1Z1);
        if (U2.eq(X)) {
            if (S2.eq(Y)) return dbl();
            return identity();
        }
        Fq H = U2.sub(X);
        Fq HH = H.sqr();
        Fq I = HH.dbl().dbl();
        Fq J = H.mul(I);
        Fq rr = S2.sub(Y).dbl();
        Fq V = X.mul(I);
        G1Jac r;
        r.X = rr.sqr().sub(J).sub(V.dbl());
        r.Y = rr.mul(V.sub(r.X)).sub(Y.mul(J).dbl());
        r.Z = Z.add(H).sqr().sub(Z1Z1).sub(HH);
        return r;
    }
};

}  // namespace rng
