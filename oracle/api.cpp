// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// C API over the oracle, consumed exclusively by tests/ (ctypes) and by
// bench.py's cpu_baseline leg.  All field elements cross this boundary as
// 4 x u64 little-endian limbs; Montgomery form unless the name says
// `canonical` — matching the reference's pinned limb layout
// (plonk_proof_def.rs:22-52).
#include <cstring>
#include <vector>
#include <omp.h>
#include "field.hpp"
#include "curve.hpp"
#include "fq2.hpp"
#include "ntt.hpp"
#include "msm.hpp"
#include "srs.hpp"
#include "keccak.hpp"
#include "transcript.hpp"
#include "plonk.hpp"
#include "poseidon2.hpp"

using namespace oracle;

extern "C" {

// ---- Fr/Fq ops (Montgomery-form in/out) ----
#define FIELD_OPS(NAME, T)                                                      \
    void orc_##NAME##_add(const u64* a, const u64* b, u64* out) {               \
        T x, y; memcpy(x.l, a, 32); memcpy(y.l, b, 32);                         \
        T r = x + y; memcpy(out, r.l, 32);                                      \
    }                                                                           \
    void orc_##NAME##_sub(const u64* a, const u64* b, u64* out) {               \
        T x, y; memcpy(x.l, a, 32); memcpy(y.l, b, 32);                         \
        T r = x - y; memcpy(out, r.l, 32);                                      \
    }                                                                           \
    void orc_##NAME##_mul(const u64* a, const u64* b, u64* out) {               \
        T x, y; memcpy(x.l, a, 32); memcpy(y.l, b, 32);                         \
        T r = x * y; memcpy(out, r.l, 32);                                      \
    }                                                                           \
    void orc_##NAME##_inv(const u64* a, u64* out) {                             \
        T x; memcpy(x.l, a, 32);                                                \
        T r = x.inverse(); memcpy(out, r.l, 32);                                \
    }                                                                           \
    void orc_##NAME##_from_canonical(const u64* a, u64* out) {                  \
        T r = T::from_canonical(a); memcpy(out, r.l, 32);                       \
    }                                                                           \
    void orc_##NAME##_to_canonical(const u64* a, u64* out) {                    \
        T x; memcpy(x.l, a, 32); x.to_canonical(out);                           \
    }

FIELD_OPS(fr, Fr)
FIELD_OPS(fq, Fq)

// ---- G1 (affine in/out: x, y Montgomery limbs + u64 infinity flag = 9 u64) ----
static void store_affine(const G1Affine& p, u64* out) {
    memcpy(out, p.x.l, 32);
    memcpy(out + 4, p.y.l, 32);
    out[8] = p.infinity ? 1 : 0;
}
static G1Affine load_affine(const u64* in) {
    G1Affine p;
    memcpy(p.x.l, in, 32);
    memcpy(p.y.l, in + 4, 32);
    p.infinity = in[8] != 0;
    return p;
}

void orc_g1_generator(u64* out) { store_affine(G1Affine::generator(), out); }

int orc_g1_is_on_curve(const u64* p) { return load_affine(p).is_on_curve() ? 1 : 0; }

void orc_g1_add(const u64* a, const u64* b, u64* out) {
    G1Proj r = G1Proj::from_affine(load_affine(a)).add_affine(load_affine(b));
    store_affine(r.to_affine(), out);
}

void orc_g1_mul(const u64* p, const u64* scalar_canonical, u64* out) {
    G1Proj r = G1Proj::from_affine(load_affine(p)).mul(scalar_canonical);
    store_affine(r.to_affine(), out);
}

// ---- NTT (in-place over n Montgomery-form Fr elements) ----
void orc_ntt(u64* data, u64 n, int inverse) {
    Fr* a = reinterpret_cast<Fr*>(data);
    if (inverse) ntt_inverse(a, n);
    else ntt_forward(a, n);
}

void orc_coset_ntt(u64* data, u64 n, const u64* g_mont, int inverse) {
    Fr* a = reinterpret_cast<Fr*>(data);
    Fr g;
    memcpy(g.l, g_mont, 32);
    if (inverse) coset_ntt_inverse(a, n, g);
    else coset_ntt_forward(a, n, g);
}

// ---- MSM ----
// bases: n * 9 u64 (affine records as above); scalars: n * 4 u64 canonical.
void orc_msm(const u64* bases, const u64* scalars, u64 n, u64* out, int window_c) {
    std::vector<G1Affine> pts(n);
    for (u64 i = 0; i < n; ++i) pts[i] = load_affine(bases + 9 * i);
    G1Proj r = msm_pippenger(pts.data(), scalars, n, window_c > 0 ? window_c : 13);
    store_affine(r.to_affine(), out);
}

void orc_msm_naive(const u64* bases, const u64* scalars, u64 n, u64* out) {
    std::vector<G1Affine> pts(n);
    for (u64 i = 0; i < n; ++i) pts[i] = load_affine(bases + 9 * i);
    G1Proj r = msm_naive(pts.data(), scalars, n);
    store_affine(r.to_affine(), out);
}

// ---- keccak256 ----
void orc_keccak256(const uint8_t* data, u64 len, uint8_t* out32) {
    keccak256(data, len, out32);
}

// ---- SRS ----
// Returns required byte length for a ptau of `power`.
u64 orc_srs_ptau_size(int power) {
    size_t npoints = (size_t(1) << power) + 3;
    return 12 + (12 + 44) + (12 + npoints * 64) + (12 + 256);
}

// Generate deterministic SRS and serialize to ptau bytes. Returns bytes written.
u64 orc_srs_generate_ptau(int power, u64 seed, uint8_t* out, u64 cap) {
    Srs srs = srs_generate(power, seed);
    std::vector<uint8_t> bytes = srs_to_ptau(srs, power);
    if (bytes.size() > cap) return 0;
    memcpy(out, bytes.data(), bytes.size());
    return bytes.size();
}

// Parse ptau bytes; writes (max_degree+1) G1 affine records; returns 0 on
// success, negative on error.
int orc_srs_parse(const uint8_t* bytes, u64 len, u64 max_degree, u64* g1_out,
                  u64* h_out /*16 u64*/, u64* beta_h_out /*16 u64*/) {
    try {
        Srs srs = parse_ptau(bytes, len, max_degree);
        for (size_t i = 0; i < srs.powers_of_g.size(); ++i)
            store_affine(srs.powers_of_g[i], g1_out + 9 * i);
        auto store_g2 = [](const G2Affine& g, u64* out) {
            memcpy(out, g.x.c0.l, 32);
            memcpy(out + 4, g.x.c1.l, 32);
            memcpy(out + 8, g.y.c0.l, 32);
            memcpy(out + 12, g.y.c1.l, 32);
        };
        store_g2(srs.h, h_out);
        store_g2(srs.beta_h, beta_h_out);
        return 0;
    } catch (...) {
        return -1;
    }
}

int orc_num_threads() { return omp_get_max_threads(); }
void orc_set_num_threads(int n) { omp_set_num_threads(n); }

// ---- TurboPlonk (oracle prover/verifier) ----
// Proof buffer layout (matches include/rng_prover.h RNG_PROOF_U64S = 157):
// 13 affine records (9 u64 each): wire comms 0..4, z comm, quot comms 0..4,
// opening, shifted opening; then 10 Fr (4 u64 Montgomery): 5 wire evals,
// 4 sigma evals, z_shift eval.

static void proof_store(const OrcProof& pf, u64* out) {
    const G1Affine* pts[13] = {&pf.wire_comms[0], &pf.wire_comms[1], &pf.wire_comms[2],
                               &pf.wire_comms[3], &pf.wire_comms[4], &pf.z_comm,
                               &pf.quot_comms[0], &pf.quot_comms[1], &pf.quot_comms[2],
                               &pf.quot_comms[3], &pf.quot_comms[4], &pf.opening,
                               &pf.shifted_opening};
    for (int i = 0; i < 13; ++i) store_affine(*pts[i], out + 9 * i);
    u64* e = out + 117;
    for (int i = 0; i < 5; ++i) memcpy(e + 4 * i, pf.wire_evals[i].l, 32);
    for (int i = 0; i < 4; ++i) memcpy(e + 20 + 4 * i, pf.sigma_evals[i].l, 32);
    memcpy(e + 36, pf.z_shift_eval.l, 32);
}

static OrcProof proof_load(const u64* in) {
    OrcProof pf;
    G1Affine* pts[13] = {&pf.wire_comms[0], &pf.wire_comms[1], &pf.wire_comms[2],
                         &pf.wire_comms[3], &pf.wire_comms[4], &pf.z_comm,
                         &pf.quot_comms[0], &pf.quot_comms[1], &pf.quot_comms[2],
                         &pf.quot_comms[3], &pf.quot_comms[4], &pf.opening,
                         &pf.shifted_opening};
    for (int i = 0; i < 13; ++i) *pts[i] = load_affine(in + 9 * i);
    const u64* e = in + 117;
    for (int i = 0; i < 5; ++i) memcpy(pf.wire_evals[i].l, e + 4 * i, 32);
    for (int i = 0; i < 4; ++i) memcpy(pf.sigma_evals[i].l, e + 20 + 4 * i, 32);
    memcpy(pf.z_shift_eval.l, e + 36, 32);
    return pf;
}

void* orc_plonk_preprocess(u64 n, u64 num_public, const u64* selectors,
                           const u64* sigma, const u64* srs_g1_records,
                           u64 srs_points) {
    OrcCircuitDesc d;
    d.n = n;
    d.num_public = num_public;
    d.selectors = reinterpret_cast<const Fr*>(selectors);
    d.sigma = sigma;
    std::vector<G1Affine> srs(srs_points);
    for (u64 i = 0; i < srs_points; ++i) srs[i] = load_affine(srs_g1_records + 9 * i);
    try {
        return orc_preprocess(d, srs);
    } catch (...) {
        return nullptr;
    }
}

void orc_plonk_pk_free(void* pk) { delete static_cast<OrcProvingKey*>(pk); }

// 18 affine records: 13 selector commitments then 5 sigma commitments —
// the same layout as the product's rng_pk_comms (canonical zeroed-infinity
// records), for direct PK parity tests
void orc_plonk_pk_comms(void* pk_, u64* out18x9) {
    auto* pk = static_cast<OrcProvingKey*>(pk_);
    auto store = [&](const G1Affine& p, u64* rec) {
        if (p.infinity) {
            memset(rec, 0, 8 * 8);
            rec[8] = 1;
            return;
        }
        store_affine(p, rec);
    };
    for (int s = 0; s < 13; ++s) store(pk->sel_comms[s], out18x9 + 9 * s);
    for (int j = 0; j < 5; ++j) store(pk->sig_comms[j], out18x9 + 9 * (13 + j));
}

int orc_plonk_prove(void* pk_, const u64* wires, const u64* pubs, u64 seed,
                    u64* out157) {
    auto* pk = static_cast<OrcProvingKey*>(pk_);
    try {
        OrcProof pf = orc_prove(*pk, reinterpret_cast<const Fr*>(wires),
                                reinterpret_cast<const Fr*>(pubs), seed);
        proof_store(pf, out157);
        return 0;
    } catch (...) {
        return -1;
    }
}

int orc_plonk_verify(void* pk_, const u64* pubs, const u64* proof157,
                     const u64* tau_canonical) {
    auto* pk = static_cast<OrcProvingKey*>(pk_);
    Fr tau = Fr::from_canonical(tau_canonical);
    try {
        OrcProof pf = proof_load(proof157);
        return orc_verify(*pk, reinterpret_cast<const Fr*>(pubs), pf, tau) ? 1 : 0;
    } catch (...) {
        return -1;
    }
}

// prove + link hint (wire-0 poly (n+2)*4 u64 + 9 u64 commitment record)
int orc_plonk_prove_with_hint(void* pk_, const u64* wires, const u64* pubs, u64 seed,
                              u64* out157, u64* out_hint) {
    auto* pk = static_cast<OrcProvingKey*>(pk_);
    try {
        std::vector<Fr> hint;
        OrcProof pf = orc_prove(*pk, reinterpret_cast<const Fr*>(wires),
                                reinterpret_cast<const Fr*>(pubs), seed, &hint);
        proof_store(pf, out157);
        memcpy(out_hint, hint.data(), hint.size() * 32);
        store_affine(pf.wire_comms[0], out_hint + 4 * hint.size());
        return 0;
    } catch (...) {
        return -1;
    }
}

// link proof between two wire-0 hints; out = 18 u64 (2 affine records)
int orc_plonk_link(void* pk_, const u64* hint_a, const u64* hint_b, u64 alignment,
                   u64 offset, u64 count, u64* out18) {
    auto* pk = static_cast<OrcProvingKey*>(pk_);
    try {
        u64 hn = pk->n + 2;
        std::vector<Fr> pa(hn), pb(hn);
        memcpy(pa.data(), hint_a, hn * 32);
        memcpy(pb.data(), hint_b, hn * 32);
        G1Affine ca = load_affine(hint_a + 4 * hn);
        G1Affine cb = load_affine(hint_b + 4 * hn);
        OrcLinkProof lp = orc_link_proofs(*pk, pa, ca, pb, cb, alignment, offset, count);
        store_affine(lp.q_comm, out18);
        store_affine(lp.opening, out18 + 9);
        return 0;
    } catch (...) {
        return -1;
    }
}

int orc_plonk_link_verify(void* pk_, const u64* comm_a9, const u64* comm_b9,
                          const u64* proof18, u64 alignment, u64 offset, u64 count,
                          const u64* tau_canonical) {
    auto* pk = static_cast<OrcProvingKey*>(pk_);
    OrcLinkProof lp;
    lp.q_comm = load_affine(proof18);
    lp.opening = load_affine(proof18 + 9);
    Fr tau = Fr::from_canonical(tau_canonical);
    return orc_link_verify(*pk, load_affine(comm_a9), load_affine(comm_b9), lp,
                           alignment, offset, count, tau)
               ? 1
               : 0;
}

// G2 generator multiple (for pairing bilinearity tests): out = 16 u64
// (x.c0, x.c1, y.c0, y.c1 Montgomery)
void orc_g2_mul_gen(const u64* scalar_canonical, u64* out16) {
    G2Affine g = G2Proj::from_affine(G2Affine::generator()).mul(scalar_canonical).to_affine();
    memcpy(out16, g.x.c0.l, 32);
    memcpy(out16 + 4, g.x.c1.l, 32);
    memcpy(out16 + 8, g.y.c0.l, 32);
    memcpy(out16 + 12, g.y.c1.l, 32);
}

// Poseidon2 hash (Montgomery limbs in/out)
void orc_poseidon2_hash(const u64* in_mont, u64 n, u64* out_mont) {
    std::vector<Fr> in(n);
    memcpy(in.data(), in_mont, n * 32);
    Fr r = poseidon2_hash(in.data(), n);
    memcpy(out_mont, r.l, 32);
}

void orc_derive_tau(u64 seed, u64* out_canonical) {
    Fr tau = derive_tau(seed);
    tau.to_canonical(out_canonical);
}

}  // extern "C"
