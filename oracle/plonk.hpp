// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// TurboPlonk KZG prover + verifier (CPU restatement).
//
// Restates the round structure of mpc-plonk's PlonkKzgSnark (non-vendored
// mpc-jellyfish@311568a), as pinned in-repo by the proof struct layout
// (types-proofs/src/rkyv_impls/plonk_proof_def.rs:197-222: 5 wire comms,
// 1 perm comm, 5 split-quotient comms, 2 opening proofs, evals {5 wires,
// 4 sigmas, 1 perm-next}) and the call sites in circuit-types/src/traits.rs
// (preprocess :850, prove :996, verify :1012).  Where the extern algebra is
// not pinned in-repo (transcript bytes, linearization grouping, blinding),
// the formulation below is the normative spec for this build — see
// transcript.hpp header and DESIGN.md §Protocol ("parity unpinned" vs the
// reference binary per SURVEY.md §8c; the enforceable contract is
// GPU == oracle bit-exact + verifier acceptance).
//
// Gate equation (reference pin: zk_gadgets/primitives/poseidon/gates.rs):
//   q_c + PI + sum q_lc_i w_i + q_m0 w0 w1 + q_m1 w2 w3 + sum q_h_i w_i^5
//       + q_ecc w0w1w2w3w4 - q_o w4 = 0   on H = <omega>, |H| = n
//
// Quotient identity:
//   gate(X) + alpha [ z(X) f(X) - z(omega X) g(X) ] + alpha^2 (z(X)-1) L1(X)
//     = t(X) Z_H(X)
//   f = prod_j (w_j + beta k_j X + gamma), g = prod_j (w_j + beta sig_j(X) + gamma)
#pragma once
#include <vector>
#include <stdexcept>
#include "field.hpp"
#include "curve.hpp"
#include "ntt.hpp"
#include "msm.hpp"
#include "transcript.hpp"

namespace oracle {

constexpr int PK_WIRES = 5;
constexpr int PK_SELS = 13;

struct OrcCircuitDesc {
    u64 n = 0, num_public = 0;
    const Fr* selectors = nullptr;   // 13 * n column-major (Montgomery)
    const u64* sigma = nullptr;      // 5 * n slot permutation indices
};

struct OrcProvingKey {
    u64 n = 0, num_public = 0;
    Fr k[PK_WIRES];                      // coset representatives (k0 = 1)
    std::vector<Fr> selq[PK_SELS];       // selector polys (coeff form, n)
    std::vector<Fr> sigp[PK_WIRES];      // sigma polys (coeff form, n)
    std::vector<Fr> sig_evals[PK_WIRES]; // sigma evals on H (k_{j'} w^{i'})
    std::vector<G1Affine> sel_comms;     // 13
    std::vector<G1Affine> sig_comms;     // 5
    std::vector<G1Affine> srs;           // commit key (max_degree+1 points)
};

// thread-local scratch for the quotient chunks between R3 and R5
inline thread_local std::vector<std::vector<Fr>> quot_chunks_tmp_;

struct OrcProof {
    G1Affine wire_comms[5];
    G1Affine z_comm;
    G1Affine quot_comms[5];
    G1Affine opening;          // W_zeta
    G1Affine shifted_opening;  // W_zeta_omega
    Fr wire_evals[5];
    Fr sigma_evals[4];
    Fr z_shift_eval;
};

// ---- small polynomial helpers (coeff vectors, ascending) ----

inline Fr poly_eval(const std::vector<Fr>& p, const Fr& x) {
    Fr acc = Fr::zero();
    for (size_t i = p.size(); i-- > 0;) acc = acc * x + p[i];
    return acc;
}

// (p - p(zeta)) / (X - zeta) via synthetic division; returns quotient.
// q[deg-1] = p[deg]; q[i-1] = p[i] + zeta*q[i]
inline std::vector<Fr> poly_div_linear(const std::vector<Fr>& p, const Fr& zeta) {
    if (p.size() < 2) return {};
    size_t d = p.size() - 1;
    std::vector<Fr> q(d, Fr::zero());
    q[d - 1] = p[d];
    for (size_t i = d - 1; i >= 1; --i) q[i - 1] = p[i] + zeta * q[i];
    return q;
}

inline void poly_add_scaled(std::vector<Fr>& acc, const std::vector<Fr>& p, const Fr& s) {
    if (acc.size() < p.size()) acc.resize(p.size(), Fr::zero());
    for (size_t i = 0; i < p.size(); ++i) acc[i] = acc[i] + p[i] * s;
}

inline G1Affine commit(const std::vector<G1Affine>& srs, const std::vector<Fr>& coeffs) {
    size_t m = coeffs.size();
    if (m > srs.size()) throw std::runtime_error("commit: poly degree exceeds SRS");
    std::vector<u64> scal(4 * m);
    for (size_t i = 0; i < m; ++i) coeffs[i].to_canonical(&scal[4 * i]);
    return msm_pippenger(srs.data(), scal.data(), m).to_affine();
}

inline void coset_ks5(Fr k[5]) {
    k[0] = Fr::one();
    Fr d = Fr::from_u64(7), acc = Fr::one();
    for (int j = 1; j < 5; ++j) {
        acc = acc * d;
        k[j] = acc;
    }
}

inline std::vector<Fr> batch_inverse(std::vector<Fr> v) {
    // Montgomery's trick
    size_t m = v.size();
    std::vector<Fr> pre(m);
    Fr acc = Fr::one();
    for (size_t i = 0; i < m; ++i) {
        pre[i] = acc;
        acc = acc * v[i];
    }
    Fr inv = acc.inverse();
    for (size_t i = m; i-- > 0;) {
        Fr orig = v[i];
        v[i] = inv * pre[i];
        inv = inv * orig;
    }
    return v;
}

// ---- preprocess (PlonkKzgSnark::preprocess, traits.rs:832-855) ----

inline OrcProvingKey* orc_preprocess(const OrcCircuitDesc& d,
                                     const std::vector<G1Affine>& srs) {
    auto* pk = new OrcProvingKey();
    pk->n = d.n;
    pk->num_public = d.num_public;
    pk->srs = srs;
    coset_ks5(pk->k);
    u64 n = d.n;

    Fr w = fr_root_of_unity(n);
    std::vector<Fr> wpow(n);
    wpow[0] = Fr::one();
    for (u64 i = 1; i < n; ++i) wpow[i] = wpow[i - 1] * w;

    for (int s = 0; s < PK_SELS; ++s) {
        pk->selq[s].assign(d.selectors + (size_t)s * n, d.selectors + (size_t)(s + 1) * n);
        ntt_inverse(pk->selq[s].data(), n);
        pk->sel_comms.push_back(commit(srs, pk->selq[s]));
    }
    for (int j = 0; j < PK_WIRES; ++j) {
        pk->sig_evals[j].resize(n);
        for (u64 i = 0; i < n; ++i) {
            u64 slot = d.sigma[(size_t)j * n + i];
            u64 jp = slot / n, ip = slot % n;
            pk->sig_evals[j][i] = pk->k[jp] * wpow[ip];
        }
        pk->sigp[j] = pk->sig_evals[j];
        ntt_inverse(pk->sigp[j].data(), n);
        pk->sig_comms.push_back(commit(srs, pk->sigp[j]));
    }
    return pk;
}

// transcript initialisation shared by prover and verifier
inline void transcript_init(Transcript& t, const OrcProvingKey& pk,
                            const Fr* pubs, u64 npub) {
    t.append_u64(pk.n);
    t.append_u64(npub);
    for (auto& c : pk.sel_comms) t.append_g1(c);
    for (auto& c : pk.sig_comms) t.append_g1(c);
    for (u64 i = 0; i < npub; ++i) t.append_fr(pubs[i]);
}

// ---- prove (PlonkKzgSnark::prove_with_link_hint, traits.rs:994-997) ----
// wires: 5*n values column-major; pubs: num_public; seed: blinder seed
// (the reference blinds from thread_rng; the fixed-seed DRBG pins bytes).
// link_hint_out (optional): receives wire-0 polynomial coefficients (n+2).

inline OrcProof orc_prove(const OrcProvingKey& pk, const Fr* wires, const Fr* pubs,
                          u64 seed, std::vector<Fr>* link_hint_out = nullptr) {
    const u64 n = pk.n;
    BlinderDrbg drbg(seed);
    Transcript tr;
    transcript_init(tr, pk, pubs, pk.num_public);

    Fr w = fr_root_of_unity(n);

    // --- R1: wire polynomials ---
    std::vector<Fr> wpoly[5];
    OrcProof pf;
    for (int j = 0; j < 5; ++j) {
        wpoly[j].assign(wires + (size_t)j * n, wires + (size_t)(j + 1) * n);
        ntt_inverse(wpoly[j].data(), n);
        Fr b0 = drbg.next(), b1 = drbg.next();
        wpoly[j].resize(n + 2, Fr::zero());
        // += (b0 + b1 X)(X^n - 1)
        wpoly[j][0] = wpoly[j][0] - b0;
        wpoly[j][1] = wpoly[j][1] - b1;
        wpoly[j][n] = wpoly[j][n] + b0;
        wpoly[j][n + 1] = wpoly[j][n + 1] + b1;
        pf.wire_comms[j] = commit(pk.srs, wpoly[j]);
        tr.append_g1(pf.wire_comms[j]);
    }
    if (link_hint_out) *link_hint_out = wpoly[0];
    Fr beta = tr.challenge();
    Fr gamma = tr.challenge();

    // --- R2: permutation grand product ---
    std::vector<Fr> znum(n), zden(n);
    {
        Fr wi = Fr::one();
        for (u64 i = 0; i < n; ++i) {
            Fr num = Fr::one(), den = Fr::one();
            for (int j = 0; j < 5; ++j) {
                Fr wv = wires[(size_t)j * n + i];
                num = num * (wv + beta * pk.k[j] * wi + gamma);
                den = den * (wv + beta * pk.sig_evals[j][i] + gamma);
            }
            znum[i] = num;
            zden[i] = den;
            wi = wi * w;
        }
    }
    std::vector<Fr> zden_inv = batch_inverse(zden);
    std::vector<Fr> zevals(n);
    zevals[0] = Fr::one();
    for (u64 i = 1; i < n; ++i) zevals[i] = zevals[i - 1] * znum[i - 1] * zden_inv[i - 1];
    std::vector<Fr> zpoly = zevals;
    ntt_inverse(zpoly.data(), n);
    {
        Fr b2 = drbg.next(), b3 = drbg.next(), b4 = drbg.next();
        zpoly.resize(n + 3, Fr::zero());
        // += (b2 X^2 + b3 X + b4)(X^n - 1)
        zpoly[0] = zpoly[0] - b4;
        zpoly[1] = zpoly[1] - b3;
        zpoly[2] = zpoly[2] - b2;
        zpoly[n] = zpoly[n] + b4;
        zpoly[n + 1] = zpoly[n + 1] + b3;
        zpoly[n + 2] = zpoly[n + 2] + b2;
    }
    pf.z_comm = commit(pk.srs, zpoly);
    tr.append_g1(pf.z_comm);
    Fr alpha = tr.challenge();

    // --- R3: quotient on coset domain of size 8n ---
    const u64 m = 8 * n;
    Fr g = Fr::from_u64(FR_GENERATOR);  // coset shift (generator, not in H)
    auto coset_evals = [&](const std::vector<Fr>& coeffs) {
        std::vector<Fr> e(m, Fr::zero());
        std::copy(coeffs.begin(), coeffs.end(), e.begin());
        coset_ntt_forward(e.data(), m, g);
        return e;
    };
    std::vector<Fr> We[5], Se[13], Ge[5], Ze, PIe, L1e;
    for (int j = 0; j < 5; ++j) We[j] = coset_evals(wpoly[j]);
    for (int s = 0; s < 13; ++s) Se[s] = coset_evals(pk.selq[s]);
    for (int j = 0; j < 5; ++j) Ge[j] = coset_evals(pk.sigp[j]);
    Ze = coset_evals(zpoly);
    {
        std::vector<Fr> pipoly(n, Fr::zero());
        for (u64 i = 0; i < pk.num_public; ++i) pipoly[i] = pubs[i];
        ntt_inverse(pipoly.data(), n);
        PIe = coset_evals(pipoly);
        std::vector<Fr> l1(n, Fr::zero());
        l1[0] = Fr::one();
        ntt_inverse(l1.data(), n);
        L1e = coset_evals(l1);
    }
    // Z_H(g x) = g^n x^n - 1 has period 8 in the coset index
    Fr gn = g.pow_u64(n);
    Fr w8 = fr_root_of_unity(m).pow_u64(n);  // primitive 8th root
    std::vector<Fr> zh8(8);
    {
        Fr cur = gn;
        for (int t = 0; t < 8; ++t) {
            zh8[t] = cur - Fr::one();
            cur = cur * w8;
        }
        zh8 = batch_inverse(zh8);
    }
    // coset x powers for the k_j beta X term
    std::vector<Fr> quot(m);
    Fr wm = fr_root_of_unity(m);
    {
        Fr x = g;
        for (u64 t = 0; t < m; ++t) {
            // gate equation
            const Fr w0 = We[0][t], w1 = We[1][t], w2 = We[2][t], w3 = We[3][t],
                     w4 = We[4][t];
            auto p5 = [](const Fr& v) {
                Fr v2 = v.square();
                return v2.square() * v;
            };
            Fr gate = Se[11][t] + PIe[t];                       // q_c + PI
            gate = gate + Se[0][t] * w0 + Se[1][t] * w1 + Se[2][t] * w2 + Se[3][t] * w3;
            gate = gate + Se[4][t] * (w0 * w1) + Se[5][t] * (w2 * w3);
            gate = gate + Se[6][t] * p5(w0) + Se[7][t] * p5(w1) + Se[8][t] * p5(w2) +
                   Se[9][t] * p5(w3);
            gate = gate + Se[12][t] * (w0 * w1 * w2 * w3 * w4);
            gate = gate - Se[10][t] * w4;
            // permutation
            Fr f = Fr::one(), gg = Fr::one();
            for (int j = 0; j < 5; ++j) {
                f = f * (We[j][t] + beta * pk.k[j] * x + gamma);
                gg = gg * (We[j][t] + beta * Ge[j][t] + gamma);
            }
            Fr zshift = Ze[(t + 8) % m];  // z(omega * coset point)
            Fr perm = alpha * (Ze[t] * f - zshift * gg);
            Fr l1term = alpha.square() * (Ze[t] - Fr::one()) * L1e[t];
            quot[t] = (gate + perm + l1term) * zh8[t % 8];
            x = x * wm;
        }
    }
    coset_ntt_inverse(quot.data(), m, g);
    // degree check: top coefficients beyond 5(n+2) must vanish
    // split into 5 chunks of (n+2) coeffs with linking blinders
    quot.resize(5 * (n + 2), Fr::zero());
    {
        Fr prev = Fr::zero();
        for (int i = 0; i < 5; ++i) {
            std::vector<Fr> chunk(quot.begin() + (size_t)i * (n + 2),
                                  quot.begin() + (size_t)(i + 1) * (n + 2));
            Fr bnext = (i < 4) ? drbg.next() : Fr::zero();
            chunk[0] = chunk[0] - prev;
            if (i < 4) {
                chunk.resize(n + 3, Fr::zero());
                chunk[n + 2] = chunk[n + 2] + bnext;
            }
            pf.quot_comms[i] = commit(pk.srs, chunk);
            tr.append_g1(pf.quot_comms[i]);
            prev = bnext;
            // store blinded chunk back for opening-phase use
            if (i == 0) quot_chunks_tmp_.clear();
            quot_chunks_tmp_.push_back(std::move(chunk));
        }
    }
    Fr zeta = tr.challenge();

    // --- R4: evaluations ---
    for (int j = 0; j < 5; ++j) {
        pf.wire_evals[j] = poly_eval(wpoly[j], zeta);
        tr.append_fr(pf.wire_evals[j]);
    }
    for (int j = 0; j < 4; ++j) {
        pf.sigma_evals[j] = poly_eval(pk.sigp[j], zeta);
        tr.append_fr(pf.sigma_evals[j]);
    }
    pf.z_shift_eval = poly_eval(zpoly, zeta * w);
    tr.append_fr(pf.z_shift_eval);
    Fr v = tr.challenge();

    // --- R5: linearization + batched openings ---
    Fr zeta_n = zeta.pow_u64(n);
    Fr zh_zeta = zeta_n - Fr::one();
    Fr l1_zeta = zh_zeta * (Fr::from_u64(n) * (zeta - Fr::one())).inverse();

    const Fr* wb = pf.wire_evals;
    auto p5 = [](const Fr& v_) {
        Fr v2 = v_.square();
        return v2.square() * v_;
    };
    // D(X) assembly in coeff form
    std::vector<Fr> D;
    poly_add_scaled(D, pk.selq[11], Fr::one());               // q_c
    for (int j = 0; j < 4; ++j) poly_add_scaled(D, pk.selq[j], wb[j]);
    poly_add_scaled(D, pk.selq[4], wb[0] * wb[1]);
    poly_add_scaled(D, pk.selq[5], wb[2] * wb[3]);
    for (int j = 0; j < 4; ++j) poly_add_scaled(D, pk.selq[6 + j], p5(wb[j]));
    poly_add_scaled(D, pk.selq[12], wb[0] * wb[1] * wb[2] * wb[3] * wb[4]);
    poly_add_scaled(D, pk.selq[10], wb[4].neg());             // -q_o w4
    Fr fbar = Fr::one(), Bbar = Fr::one();
    for (int j = 0; j < 5; ++j) fbar = fbar * (wb[j] + beta * pk.k[j] * zeta + gamma);
    for (int j = 0; j < 4; ++j) Bbar = Bbar * (wb[j] + beta * pf.sigma_evals[j] + gamma);
    poly_add_scaled(D, zpoly, alpha * fbar + alpha.square() * l1_zeta);
    poly_add_scaled(D, pk.sigp[4], (alpha * beta * pf.z_shift_eval * Bbar).neg());
    {
        Fr zpow = zh_zeta.neg();  // -Z_H(zeta) * zeta^{i(n+2)}
        Fr step = zeta.pow_u64(n + 2);
        for (int i = 0; i < 5; ++i) {
            poly_add_scaled(D, quot_chunks_tmp_[i], zpow);
            zpow = zpow * step;
        }
    }

    // combined opening at zeta: C = D + sum v^i P_i  (P = w0..w4, sig0..3)
    std::vector<Fr> C = D;
    Fr vp = Fr::one();
    for (int j = 0; j < 5; ++j) {
        vp = vp * v;
        poly_add_scaled(C, wpoly[j], vp);
    }
    for (int j = 0; j < 4; ++j) {
        vp = vp * v;
        poly_add_scaled(C, pk.sigp[j], vp);
    }
    std::vector<Fr> Wz = poly_div_linear(C, zeta);
    pf.opening = commit(pk.srs, Wz);
    std::vector<Fr> Wzw = poly_div_linear(zpoly, zeta * w);
    pf.shifted_opening = commit(pk.srs, Wzw);
    tr.append_g1(pf.opening);
    tr.append_g1(pf.shifted_opening);
    return pf;
}

// ---- verify ----
// The oracle verifier uses the SRS trapdoor tau (our SRS is generated with a
// known seed — SURVEY.md §0.4): the KZG pairing check
//   e(W + u W', [tau]_2) == e(zeta W + u zeta omega W' + F - E G, [1]_2)
// is equivalent (H of prime order, honest SRS) to the G1 equation
//   tau (W + u W') == zeta W + u zeta omega W' + F - E G.
// The PRODUCT side (rng_verify) performs the real pairing; the oracle stays
// pairing-free by design.

inline bool orc_verify(const OrcProvingKey& pk, const Fr* pubs, const OrcProof& pf,
                       const Fr& tau) {
    const u64 n = pk.n;
    Fr w = fr_root_of_unity(n);
    Transcript tr;
    transcript_init(tr, pk, pubs, pk.num_public);
    for (int j = 0; j < 5; ++j) tr.append_g1(pf.wire_comms[j]);
    Fr beta = tr.challenge();
    Fr gamma = tr.challenge();
    tr.append_g1(pf.z_comm);
    Fr alpha = tr.challenge();
    for (int i = 0; i < 5; ++i) tr.append_g1(pf.quot_comms[i]);
    Fr zeta = tr.challenge();
    for (int j = 0; j < 5; ++j) tr.append_fr(pf.wire_evals[j]);
    for (int j = 0; j < 4; ++j) tr.append_fr(pf.sigma_evals[j]);
    tr.append_fr(pf.z_shift_eval);
    Fr v = tr.challenge();
    tr.append_g1(pf.opening);
    tr.append_g1(pf.shifted_opening);
    Fr u = tr.challenge();

    Fr zeta_n = zeta.pow_u64(n);
    Fr zh_zeta = zeta_n - Fr::one();
    Fr l1_zeta = zh_zeta * (Fr::from_u64(n) * (zeta - Fr::one())).inverse();
    // PI(zeta) = sum pub_i L_i(zeta), L_i(zeta) = w^i zh / (n (zeta - w^i))
    Fr pi_zeta = Fr::zero();
    {
        std::vector<Fr> dens(pk.num_public);
        Fr wi = Fr::one();
        std::vector<Fr> wis(pk.num_public);
        for (u64 i = 0; i < pk.num_public; ++i) {
            wis[i] = wi;
            dens[i] = Fr::from_u64(n) * (zeta - wi);
            wi = wi * w;
        }
        if (pk.num_public) {
            dens = batch_inverse(dens);
            for (u64 i = 0; i < pk.num_public; ++i)
                pi_zeta = pi_zeta + pubs[i] * wis[i] * zh_zeta * dens[i];
        }
    }

    const Fr* wb = pf.wire_evals;
    auto p5 = [](const Fr& x) {
        Fr x2 = x.square();
        return x2.square() * x;
    };
    Fr fbar = Fr::one(), Bbar = Fr::one();
    for (int j = 0; j < 5; ++j) fbar = fbar * (wb[j] + beta * pk.k[j] * zeta + gamma);
    for (int j = 0; j < 4; ++j) Bbar = Bbar * (wb[j] + beta * pf.sigma_evals[j] + gamma);
    Fr ED = pi_zeta.neg() + alpha * pf.z_shift_eval * Bbar * (wb[4] + gamma) +
            alpha.square() * l1_zeta;

    // [D] as G1 linear combination
    G1Proj Dc = G1Proj::identity();
    auto addc = [&](const G1Affine& c, const Fr& s) {
        u64 sc[4];
        s.to_canonical(sc);
        Dc = Dc.add(G1Proj::from_affine(c).mul(sc));
    };
    addc(pk.sel_comms[11], Fr::one());
    for (int j = 0; j < 4; ++j) addc(pk.sel_comms[j], wb[j]);
    addc(pk.sel_comms[4], wb[0] * wb[1]);
    addc(pk.sel_comms[5], wb[2] * wb[3]);
    for (int j = 0; j < 4; ++j) addc(pk.sel_comms[6 + j], p5(wb[j]));
    addc(pk.sel_comms[12], wb[0] * wb[1] * wb[2] * wb[3] * wb[4]);
    addc(pk.sel_comms[10], wb[4].neg());
    addc(pf.z_comm, alpha * fbar + alpha.square() * l1_zeta);
    addc(pk.sig_comms[4], (alpha * beta * pf.z_shift_eval * Bbar).neg());
    {
        Fr zpow = zh_zeta.neg();
        Fr step = zeta.pow_u64(n + 2);
        for (int i = 0; i < 5; ++i) {
            addc(pf.quot_comms[i], zpow);
            zpow = zpow * step;
        }
    }

    // F = [D] + sum v^i [P_i] + u [z];  E = ED + sum v^i pbar_i + u zbar_w
    G1Proj F = Dc;
    Fr E = ED;
    Fr vp = Fr::one();
    for (int j = 0; j < 5; ++j) {
        vp = vp * v;
        u64 sc[4];
        vp.to_canonical(sc);
        F = F.add(G1Proj::from_affine(pf.wire_comms[j]).mul(sc));
        E = E + vp * pf.wire_evals[j];
    }
    for (int j = 0; j < 4; ++j) {
        vp = vp * v;
        u64 sc[4];
        vp.to_canonical(sc);
        F = F.add(G1Proj::from_affine(pk.sig_comms[j]).mul(sc));
        E = E + vp * pf.sigma_evals[j];
    }
    {
        u64 sc[4];
        u.to_canonical(sc);
        F = F.add(G1Proj::from_affine(pf.z_comm).mul(sc));
        E = E + u * pf.z_shift_eval;
    }

    // tau (W + u W') == zeta W + u zeta w W' + F - E G
    G1Proj Wz = G1Proj::from_affine(pf.opening);
    G1Proj Wzw = G1Proj::from_affine(pf.shifted_opening);
    u64 sc[4];
    u.to_canonical(sc);
    G1Proj lhs_in = Wz.add(Wzw.mul(sc));
    Fr tau_f = tau;
    u64 tc[4];
    tau_f.to_canonical(tc);
    G1Proj lhs = lhs_in.mul(tc);

    G1Proj rhs = G1Proj::identity();
    {
        u64 zc[4];
        zeta.to_canonical(zc);
        rhs = rhs.add(Wz.mul(zc));
        Fr uzw = u * zeta * w;
        uzw.to_canonical(zc);
        rhs = rhs.add(Wzw.mul(zc));
        rhs = rhs.add(F);
        Fr En = E.neg();
        En.to_canonical(zc);
        rhs = rhs.add(G1Proj::from_affine(G1Affine::generator()).mul(zc));
    }
    return lhs.to_affine() == rhs.to_affine();
}

// ---- proof linking (PlonkKzgSnark::link_proofs, called at
// circuits-core/src/zk_circuits/proof_linking/intent_and_balance.rs:66-73;
// hint struct pinned at plonk_proof_def.rs:143-150) ----
//
// Spec for this build: for link-group positions
// S = { w^i : i in [offset, offset+count) } shared by two proofs' wire-0
// polynomials a(X), b(X):
//   q(X) = (a(X) - b(X)) / Z_S(X)            (exact iff they agree on S)
//   transcript: append [a], [b], [q] -> challenge eta
//   W(X) = F(X) / (X - eta),  F(X) = a(X) - b(X) - Z_S(eta) q(X)  (F(eta)=0)
// link proof = ([q], [W]).  Verify: F-commitment opens to 0 at eta.

struct OrcLinkProof {
    G1Affine q_comm, opening;
};

// positions live on the shared alignment grid H_{2^a}: x_k = w_{2^a}^(off+k)
inline std::vector<Fr> vanishing_of_positions(u64 alignment, u64 offset, u64 count) {
    Fr w = fr_root_of_unity(u64(1) << alignment);
    Fr wi = w.pow_u64(offset);
    std::vector<Fr> z{Fr::one()};
    for (u64 i = 0; i < count; ++i) {
        // multiply by (X - w^(offset+i))
        std::vector<Fr> nz(z.size() + 1, Fr::zero());
        for (size_t j = 0; j < z.size(); ++j) {
            nz[j + 1] = nz[j + 1] + z[j];
            nz[j] = nz[j] - z[j] * wi;
        }
        z = std::move(nz);
        wi = wi * w;
    }
    return z;
}

// long division p / d (d monic); returns quotient, ignores remainder
inline std::vector<Fr> poly_div(const std::vector<Fr>& p, const std::vector<Fr>& d) {
    if (p.size() < d.size()) return {Fr::zero()};
    std::vector<Fr> r = p;
    size_t dd = d.size() - 1;
    std::vector<Fr> q(p.size() - dd, Fr::zero());
    for (size_t i = p.size(); i-- > dd;) {
        Fr c = r[i];  // d is monic
        q[i - dd] = c;
        if (c.is_zero()) continue;
        for (size_t j = 0; j <= dd; ++j) r[i - dd + j] = r[i - dd + j] - d[j] * c;
    }
    return q;
}

inline OrcLinkProof orc_link_proofs(const OrcProvingKey& pk, const std::vector<Fr>& poly_a,
                                    const G1Affine& comm_a, const std::vector<Fr>& poly_b,
                                    const G1Affine& comm_b, u64 alignment, u64 offset,
                                    u64 count) {
    std::vector<Fr> diff = poly_a;
    {
        if (diff.size() < poly_b.size()) diff.resize(poly_b.size(), Fr::zero());
        for (size_t i = 0; i < poly_b.size(); ++i) diff[i] = diff[i] - poly_b[i];
    }
    std::vector<Fr> zs = vanishing_of_positions(alignment, offset, count);
    std::vector<Fr> q = poly_div(diff, zs);
    OrcLinkProof lp;
    lp.q_comm = commit(pk.srs, q);
    Transcript tr;
    tr.append_u64(alignment);
    tr.append_u64(offset);
    tr.append_u64(count);
    tr.append_g1(comm_a);
    tr.append_g1(comm_b);
    tr.append_g1(lp.q_comm);
    Fr eta = tr.challenge();
    Fr zs_eta = poly_eval(zs, eta);
    std::vector<Fr> F = diff;
    poly_add_scaled(F, q, zs_eta.neg());
    std::vector<Fr> W = poly_div_linear(F, eta);
    lp.opening = commit(pk.srs, W);
    return lp;
}

inline bool orc_link_verify(const OrcProvingKey& pk, const G1Affine& comm_a,
                            const G1Affine& comm_b, const OrcLinkProof& lp,
                            u64 alignment, u64 offset, u64 count, const Fr& tau) {
    Transcript tr;
    tr.append_u64(alignment);
    tr.append_u64(offset);
    tr.append_u64(count);
    tr.append_g1(comm_a);
    tr.append_g1(comm_b);
    tr.append_g1(lp.q_comm);
    Fr eta = tr.challenge();
    std::vector<Fr> zs = vanishing_of_positions(alignment, offset, count);
    Fr zs_eta = poly_eval(zs, eta);
    // F = A - B - zs_eta * Q ; check tau*W == eta*W + F
    G1Proj F = G1Proj::from_affine(comm_a)
                   .add(G1Proj::from_affine(comm_b).neg());
    u64 sc[4];
    zs_eta.neg().to_canonical(sc);
    F = F.add(G1Proj::from_affine(lp.q_comm).mul(sc));
    G1Proj W = G1Proj::from_affine(lp.opening);
    u64 tc[4];
    tau.to_canonical(tc);
    G1Proj lhs = W.mul(tc);
    u64 ec[4];
    eta.to_canonical(ec);
    G1Proj rhs = W.mul(ec).add(F);
    return lhs.to_affine() == rhs.to_affine();
}

}  // namespace oracle
