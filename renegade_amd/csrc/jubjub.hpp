// Baby Jubjub (EIP-2494) host arithmetic + Schnorr + ElGamal-hybrid natives.
//
// The reference's embedded curve (constants/src/lib.rs:73-85,
// circuit-types/src/primitives/{baby_jubjub,schnorr,elgamal}.rs).  The
// signature / encryption schemes here keep the reference's SHAPE
// (schnorr.rs:43-54 "s = H(M || r) * sk + k"; elgamal.rs hybrid pad stream
// seeded by the shared point) but use Poseidon2 as the challenge / pad hash:
// the reference delegates both to jf-primitives 0.4.0-pre.0
// (mpc-jellyfish @311568a, NOT vendored in /root/reference), whose Rescue
// parameterization cannot be pinned offline — documented in DESIGN.md
// ("parity unpinned" for the embedded-curve hash choice).
#pragma once
#include "gpu_field.hpp"
#include "poseidon2.hpp"
#include "../../include/babyjubjub_params.h"

namespace rng {

struct JjPoint {  // affine twisted Edwards, identity = (0, 1)
    Fr x, y;
    static JjPoint identity() { return {Fr::zero(), Fr::one()}; }
    bool is_identity() const { return x.is_zero() && witness_eq(y, Fr::one()); }
    static bool witness_eq(const Fr& a, const Fr& b) {
        u64 la[4], lb[4];
        a.to_canonical(la);
        b.to_canonical(lb);
        return la[0] == lb[0] && la[1] == lb[1] && la[2] == lb[2] && la[3] == lb[3];
    }
};

inline Fr jj_a() {
    static const u64 l[4] = JJ_A_MONT;
    Fr r;
    r.l[0] = l[0], r.l[1] = l[1], r.l[2] = l[2], r.l[3] = l[3];
    return r;
}
inline Fr jj_d() {
    static const u64 l[4] = JJ_D_MONT;
    Fr r;
    r.l[0] = l[0], r.l[1] = l[1], r.l[2] = l[2], r.l[3] = l[3];
    return r;
}
inline JjPoint jj_base() {
    static const u64 lx[4] = JJ_BASE_X_MONT;
    static const u64 ly[4] = JJ_BASE_Y_MONT;
    JjPoint p;
    p.x.l[0] = lx[0], p.x.l[1] = lx[1], p.x.l[2] = lx[2], p.x.l[3] = lx[3];
    p.y.l[0] = ly[0], p.y.l[1] = ly[1], p.y.l[2] = ly[2], p.y.l[3] = ly[3];
    return p;
}

inline bool jj_on_curve(const JjPoint& p) {
    // a*x^2 + y^2 == 1 + d*x^2*y^2
    Fr x2 = p.x.sqr(), y2 = p.y.sqr();
    Fr lhs = jj_a().mul(x2).add(y2);
    Fr rhs = Fr::one().add(jj_d().mul(x2).mul(y2));
    return JjPoint::witness_eq(lhs, rhs);
}

// complete twisted Edwards addition (a QR, d non-QR: no exceptional cases)
inline JjPoint jj_add(const JjPoint& p, const JjPoint& q) {
    Fr x1x2 = p.x.mul(q.x), y1y2 = p.y.mul(q.y);
    Fr x1y2 = p.x.mul(q.y), y1x2 = p.y.mul(q.x);
    Fr t = jj_d().mul(x1x2).mul(y1y2);
    JjPoint r;
    r.x = x1y2.add(y1x2).mul(Fr::one().add(t).inverse());
    r.y = y1y2.sub(jj_a().mul(x1x2)).mul(Fr::one().sub(t).inverse());
    return r;
}

// scalar (plain LE 4-limb) * point, double-and-add
inline JjPoint jj_mul(const u64 k[4], const JjPoint& p) {
    JjPoint acc = JjPoint::identity();
    JjPoint add_ = p;
    for (int i = 0; i < 256; ++i) {
        if ((k[i / 64] >> (i % 64)) & 1) acc = jj_add(acc, add_);
        add_ = jj_add(add_, add_);
    }
    return acc;
}

// ---- scalar arithmetic mod l (plain form; used only host-side, rarely) ----
struct JjScalar {
    u64 v[4];  // plain little-endian, < l
};

inline bool jj_sc_gte(const u64 a[4], const u64 b[4]) {
    for (int i = 3; i >= 0; --i) {
        if (a[i] != b[i]) return a[i] > b[i];
    }
    return true;
}
inline void jj_sc_sub(u64 a[4], const u64 b[4]) {
    unsigned __int128 borrow = 0;
    for (int i = 0; i < 4; ++i) {
        unsigned __int128 t = (unsigned __int128)a[i] - b[i] - (u64)borrow;
        a[i] = (u64)t;
        borrow = (t >> 64) ? 1 : 0;
    }
}
inline JjScalar jj_sc_add(const JjScalar& a, const JjScalar& b) {
    static const u64 L[4] = JJ_ORDER_PLAIN;
    JjScalar r;
    unsigned __int128 carry = 0;
    for (int i = 0; i < 4; ++i) {
        unsigned __int128 t = (unsigned __int128)a.v[i] + b.v[i] + (u64)carry;
        r.v[i] = (u64)t;
        carry = t >> 64;
    }
    if (carry || jj_sc_gte(r.v, L)) jj_sc_sub(r.v, L);
    return r;
}
inline JjScalar jj_sc_dbl(const JjScalar& a) { return jj_sc_add(a, a); }
// mul via double-and-add on 251 bits (host-only, few calls per signature)
inline JjScalar jj_sc_mul(const JjScalar& a, const JjScalar& b) {
    JjScalar acc{{0, 0, 0, 0}};
    for (int i = JJ_ORDER_BITS; i >= 0; --i) {
        acc = jj_sc_dbl(acc);
        if ((b.v[i / 64] >> (i % 64)) & 1) acc = jj_sc_add(acc, a);
    }
    return acc;
}

// ---- Schnorr (schnorr.rs shape; Poseidon2 challenge, see header note) ----
// challenge c = low 248 bits of Poseidon2(vk.x, vk.y, R.x, R.y, msg...)
inline JjScalar jj_challenge(const JjPoint& vk, const JjPoint& R, const Fr* msg,
                             size_t n) {
    std::vector<Fr> in = {vk.x, vk.y, R.x, R.y};
    in.insert(in.end(), msg, msg + n);
    Fr h = poseidon_hash(in.data(), in.size());
    JjScalar c;
    h.to_canonical(c.v);
    c.v[3] &= (1ull << 56) - 1;  // keep 248 bits < l
    return c;
}

struct JjSignature {
    JjScalar s;
    JjPoint R;
};

inline JjPoint jj_pubkey(const JjScalar& sk) { return jj_mul(sk.v, jj_base()); }

inline JjSignature jj_sign(const JjScalar& sk, const JjScalar& k, const Fr* msg,
                           size_t n) {
    JjSignature sig;
    sig.R = jj_mul(k.v, jj_base());
    JjScalar c = jj_challenge(jj_pubkey(sk), sig.R, msg, n);
    sig.s = jj_sc_add(jj_sc_mul(c, sk), k);  // s = c*sk + k
    return sig;
}

inline bool jj_verify(const JjPoint& vk, const JjSignature& sig, const Fr* msg,
                      size_t n) {
    JjScalar c = jj_challenge(vk, sig.R, msg, n);
    JjPoint lhs = jj_mul(sig.s.v, jj_base());
    JjPoint rhs = jj_add(sig.R, jj_mul(c.v, vk));
    return JjPoint::witness_eq(lhs.x, rhs.x) && JjPoint::witness_eq(lhs.y, rhs.y);
}

// ---- ElGamal hybrid (elgamal.rs shape; Poseidon2 pad stream) ----
// eph = k*B; shared = k*pk; pad_i = Poseidon2(shared.x, shared.y, i);
// c_i = m_i + pad_i
template <int N>
struct JjCiphertext {
    JjPoint ephemeral_key;
    Fr ciphertext[N];
};

template <int N>
inline JjCiphertext<N> jj_elgamal_encrypt(const JjPoint& pk, const JjScalar& k,
                                          const Fr* msg) {
    JjCiphertext<N> ct;
    ct.ephemeral_key = jj_mul(k.v, jj_base());
    JjPoint shared = jj_mul(k.v, pk);
    for (int i = 0; i < N; ++i) {
        Fr in[3] = {shared.x, shared.y, Fr::from_u64((u64)i)};
        ct.ciphertext[i] = msg[i].add(poseidon_hash(in, 3));
    }
    return ct;
}

}  // namespace rng
