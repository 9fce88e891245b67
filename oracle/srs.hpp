// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Deterministic powers-of-tau SRS in snarkjs .ptau layout + parser.
//
// Restates the parser semantics of
//   crates/circuits/circuit-types/src/primitives/srs.rs:63-214
// byte-for-byte: magic "ptau", version 1, 11 sections; section headers are
// (u32 section_num, u64 section_size); section 1 = (u32 n8, modulus LE bytes,
// u32 power, u32 ceremony_power); section 2 = G1 points as raw Montgomery
// 4 x u64 LE (x, y); section 3 = two G2 points (x.c0, x.c1, y.c0, y.c1).
// The repo's own SRS binary is ABSENT from the snapshot (srs.rs:33-35
// includes files that do not exist — SURVEY.md §0.4), so the SRS here is
// GENERATED with a known tau derived from a seed; the parser surface and the
// pairing-ratio well-formedness test (srs.rs:236-266) are what is preserved.
// Generation writes only the points the parser reads (MAX_SRS_DEGREE+1 G1,
// 2 G2) with section sizes set accordingly; snarkjs writes more points per
// section, which the reference parser skips via the size field anyway.
#pragma once
#include <vector>
#include <string>
#include <stdexcept>
#include <omp.h>
#include "curve.hpp"
#include "fq2.hpp"
#include "keccak.hpp"

namespace oracle {

constexpr int MAX_SRS_POWER = 17;                        // srs.rs:45
constexpr size_t MAX_SRS_DEGREE = (1u << MAX_SRS_POWER) + 2;  // srs.rs:47

struct Srs {
    std::vector<G1Affine> powers_of_g;  // tau^i * G, i = 0..=degree
    G2Affine h;                         // H
    G2Affine beta_h;                    // tau * H
};

// Derive tau from a seed: tau = keccak256("renegade-amd-srs-tau" || le64(seed)) mod r.
inline Fr derive_tau(u64 seed) {
    uint8_t buf[32 + 8];
    const char* tag = "renegade-amd-srs-tau";
    uint8_t msg[28];
    memcpy(msg, tag, 20);
    memcpy(msg + 20, &seed, 8);
    keccak256(msg, 28, buf);
    // interpret as LE 256-bit, reduce mod r via from_canonical on a value < 2^256
    // (from_canonical multiplies by R2 which performs full reduction)
    u64 limbs[4];
    memcpy(limbs, buf, 32);
    return Fr::from_canonical(limbs);
}

// Generate tau-power G1 points with a fixed-base table (G_j = 2^j * G).
inline Srs srs_generate(int power, u64 seed) {
    size_t degree = (size_t(1) << power) + 2;
    Fr tau = derive_tau(seed);

    // scalar powers tau^i (canonical form) -- sequential, cheap
    std::vector<std::array<u64, 4>> scalars(degree + 1);
    Fr acc = Fr::one();
    for (size_t i = 0; i <= degree; ++i) {
        acc.to_canonical(scalars[i].data());
        acc = acc * tau;
    }
    // recompute: scalars[i] should be tau^i; fix order (acc started at 1)
    // (loop above already stores tau^i then multiplies)

    // fixed-base table
    std::vector<G1Proj> table(256);
    table[0] = G1Proj::from_affine(G1Affine::generator());
    for (int j = 1; j < 256; ++j) table[j] = table[j - 1].dbl();

    std::vector<G1Affine> points(degree + 1);
#pragma omp parallel for schedule(static)
    for (long long i = 0; i <= (long long)degree; ++i) {
        G1Proj p = G1Proj::identity();
        const u64* s = scalars[i].data();
        for (int b = 0; b < 256; ++b)
            if ((s[b / 64] >> (b % 64)) & 1) p = p.add(table[b]);
        points[i] = p.to_affine();
    }

    Srs srs;
    srs.powers_of_g = std::move(points);
    srs.h = G2Affine::generator();
    u64 tau_c[4];
    tau.to_canonical(tau_c);
    srs.beta_h = G2Proj::from_affine(srs.h).mul(tau_c).to_affine();
    return srs;
}

// ---- ptau serialization (layout per srs.rs:63-214) ----

namespace detail {
inline void put_u32(std::vector<uint8_t>& v, uint32_t x) {
    v.insert(v.end(), (uint8_t*)&x, (uint8_t*)&x + 4);
}
inline void put_u64(std::vector<uint8_t>& v, uint64_t x) {
    v.insert(v.end(), (uint8_t*)&x, (uint8_t*)&x + 8);
}
inline void put_fq(std::vector<uint8_t>& v, const Fq& f) {
    v.insert(v.end(), (const uint8_t*)f.l, (const uint8_t*)f.l + 32);  // raw Montgomery limbs
}
}  // namespace detail

inline std::vector<uint8_t> srs_to_ptau(const Srs& srs, int power) {
    using namespace detail;
    std::vector<uint8_t> out;
    out.insert(out.end(), {'p', 't', 'a', 'u'});
    put_u32(out, 1);   // version (srs.rs:54)
    put_u32(out, 11);  // num sections (srs.rs:56)

    // section 1: header
    put_u32(out, 1);
    uint64_t s1_size = 4 + 32 + 4 + 4;
    put_u64(out, s1_size);
    put_u32(out, 32);  // modulus byte length
    static constexpr u64 qmod[4] = FQ_MODULUS;
    out.insert(out.end(), (const uint8_t*)qmod, (const uint8_t*)qmod + 32);
    put_u32(out, (uint32_t)power);
    put_u32(out, (uint32_t)power);  // ceremony power

    // section 2: G1 points
    put_u32(out, 2);
    uint64_t npoints = srs.powers_of_g.size();
    put_u64(out, npoints * 64);
    for (const auto& p : srs.powers_of_g) {
        put_fq(out, p.x);
        put_fq(out, p.y);
    }

    // section 3: G2 points h, beta_h
    put_u32(out, 3);
    put_u64(out, 2 * 128);
    for (const G2Affine* g : {&srs.h, &srs.beta_h}) {
        put_fq(out, g->x.c0);
        put_fq(out, g->x.c1);
        put_fq(out, g->y.c0);
        put_fq(out, g->y.c1);
    }
    return out;
}

// Parser mirroring parse_ptau_file (srs.rs:63-214), including its asserts.
inline Srs parse_ptau(const uint8_t* bytes, size_t len, size_t max_degree = MAX_SRS_DEGREE) {
    size_t pos = 0;
    auto need = [&](size_t n) {
        if (pos + n > len) throw std::runtime_error("ptau: truncated");
    };
    auto rd_u32 = [&]() { need(4); uint32_t v; memcpy(&v, bytes + pos, 4); pos += 4; return v; };
    auto rd_u64 = [&]() { need(8); uint64_t v; memcpy(&v, bytes + pos, 8); pos += 8; return v; };
    auto rd_fq = [&]() { need(32); Fq f; memcpy(f.l, bytes + pos, 32); pos += 32; return f; };

    need(4);
    if (memcmp(bytes, "ptau", 4) != 0) throw std::runtime_error("ptau: bad magic");
    pos += 4;
    if (rd_u32() != 1) throw std::runtime_error("ptau: bad version");
    if (rd_u32() != 11) throw std::runtime_error("ptau: bad num sections");

    // section 1
    if (rd_u32() != 1) throw std::runtime_error("ptau: bad section 1 num");
    uint64_t s1_size = rd_u64();
    size_t s1_end = pos + s1_size;
    uint32_t n8 = rd_u32();
    if (n8 != 32) throw std::runtime_error("ptau: modulus size != 32");
    static constexpr u64 qmod[4] = FQ_MODULUS;
    need(32);
    if (memcmp(bytes + pos, qmod, 32) != 0) throw std::runtime_error("ptau: modulus mismatch");
    pos += 32;
    uint32_t power = rd_u32();
    (void)rd_u32();  // ceremony power
    if (power < MAX_SRS_POWER && max_degree == MAX_SRS_DEGREE)
        throw std::runtime_error("ptau: power too small");
    pos = s1_end;

    Srs srs;
    // section 2
    if (rd_u32() != 2) throw std::runtime_error("ptau: bad section 2 num");
    uint64_t s2_size = rd_u64();
    size_t s2_end = pos + s2_size;
    srs.powers_of_g.reserve(max_degree + 1);
    for (size_t i = 0; i <= max_degree; ++i) {
        G1Affine p;
        p.x = rd_fq();
        p.y = rd_fq();
        p.infinity = false;
        if (!p.is_on_curve()) throw std::runtime_error("ptau: G1 point not on curve");
        srs.powers_of_g.push_back(p);
    }
    pos = s2_end;

    // section 3
    if (rd_u32() != 3) throw std::runtime_error("ptau: bad section 3 num");
    (void)rd_u64();
    for (G2Affine* g : {&srs.h, &srs.beta_h}) {
        g->x.c0 = rd_fq();
        g->x.c1 = rd_fq();
        g->y.c0 = rd_fq();
        g->y.c1 = rd_fq();
        g->infinity = false;
        if (!g->is_on_curve()) throw std::runtime_error("ptau: G2 point not on curve");
    }
    return srs;
}

}  // namespace oracle
