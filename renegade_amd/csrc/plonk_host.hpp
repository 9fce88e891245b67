// PRODUCT PATH — TurboPlonk prover host orchestration (MI355X-native).
//
// Replaces mpc-plonk's PlonkKzgSnark::{preprocess, prove_with_link_hint}
// (called at circuit-types/src/traits.rs:850,996).  All NTT/MSM/quotient
// work runs on the GPU (kernels in ntt_kernels.hip / msm_kernels.hip /
// the quotient kernel below); the host does transcripts, blinding, the
// O(n) linearization/opening assembly, and the tiny EC folds.
//
// Transcript / blinder-DRBG byte spec: identical to the oracle's
// (oracle/transcript.hpp header comment is the normative spec); implemented
// here independently — the product never links oracle code.
#pragma once
#include <vector>
#include <cstring>
#include "gpu_field.hpp"
#include "gpu_curve.hpp"
#include "plonk_circuit.hpp"

namespace rng {

// ---- keccak-256 (independent implementation; pinned by known-answer
// vectors in tests) ----
struct Keccak {
    uint64_t a[25] = {0};
    uint8_t buf[136];
    size_t fill = 0;

    // n == 0 must not shift by 64 (UB clang -O3 miscompiles)
    static uint64_t rol(uint64_t x, int n) {
        return n == 0 ? x : (x << n) | (x >> (64 - n));
    }

    void f1600() {
        static const uint64_t RC[24] = {
            0x0000000000000001ULL, 0x0000000000008082ULL, 0x800000000000808aULL,
            0x8000000080008000ULL, 0x000000000000808bULL, 0x0000000080000001ULL,
            0x8000000080008081ULL, 0x8000000000008009ULL, 0x000000000000008aULL,
            0x0000000000000088ULL, 0x0000000080008009ULL, 0x000000008000000aULL,
            0x000000008000808bULL, 0x800000000000008bULL, 0x8000000000008089ULL,
            0x8000000000008003ULL, 0x8000000000008002ULL, 0x8000000000000080ULL,
            0x000000000000800aULL, 0x800000008000000aULL, 0x8000000080008081ULL,
            0x8000000000008080ULL, 0x0000000080000001ULL, 0x8000000080008008ULL};
        // rotation offsets indexed by linear position x+5y
        static const int ROT[25] = {0,  1,  62, 28, 27, 36, 44, 6,  55, 20, 3, 10, 43,
                                    25, 39, 41, 45, 15, 21, 8,  18, 2,  61, 56, 14};
        for (int rnd = 0; rnd < 24; ++rnd) {
            uint64_t c[5], d[5];
            for (int x = 0; x < 5; ++x)
                c[x] = a[x] ^ a[x + 5] ^ a[x + 10] ^ a[x + 15] ^ a[x + 20];
            for (int x = 0; x < 5; ++x) d[x] = c[(x + 4) % 5] ^ rol(c[(x + 1) % 5], 1);
            uint64_t b[25];
            for (int y = 0; y < 5; ++y)
                for (int x = 0; x < 5; ++x) {
                    uint64_t v = a[x + 5 * y] ^ d[x];
                    int nx = y, ny = (2 * x + 3 * y) % 5;
                    b[nx + 5 * ny] = rol(v, ROT[x + 5 * y]);
                }
            for (int y = 0; y < 5; ++y)
                for (int x = 0; x < 5; ++x)
                    a[x + 5 * y] =
                        b[x + 5 * y] ^ ((~b[(x + 1) % 5 + 5 * y]) & b[(x + 2) % 5 + 5 * y]);
            a[0] ^= RC[rnd];
        }
    }
    void update(const uint8_t* p, size_t len) {
        while (len) {
            size_t take = 136 - fill;
            if (take > len) take = len;
            memcpy(buf + fill, p, take);
            fill += take;
            p += take;
            len -= take;
            if (fill == 136) {
                for (int i = 0; i < 17; ++i) {
                    uint64_t v;
                    memcpy(&v, buf + 8 * i, 8);
                    a[i] ^= v;
                }
                f1600();
                fill = 0;
            }
        }
    }
    void digest(uint8_t out[32]) {
        uint8_t blk[136];
        memcpy(blk, buf, fill);
        memset(blk + fill, 0, 136 - fill);
        blk[fill] ^= 0x01;
        blk[135] ^= 0x80;
        for (int i = 0; i < 17; ++i) {
            uint64_t v;
            memcpy(&v, blk + 8 * i, 8);
            a[i] ^= v;
        }
        f1600();
        memcpy(out, a, 32);
    }
};

inline void keccak256_h(const uint8_t* p, size_t n, uint8_t out[32]) {
    Keccak k;
    k.update(p, n);
    k.digest(out);
}

inline Fr fr_from_hash(const uint8_t h[32]) {
    u64 l[4];
    memcpy(l, h, 32);
    return Fr::from_canonical(l);  // LE int mod r (full reduction via R2 mul)
}

// ---- transcript (spec: oracle/transcript.hpp header) ----
struct HostTranscript {
    uint8_t state[32] = {0};
    std::vector<uint8_t> buf;
    void ab(const void* p, size_t n) {
        buf.insert(buf.end(), (const uint8_t*)p, (const uint8_t*)p + n);
    }
    void append_u64(uint64_t x) { ab(&x, 8); }
    void append_fr(const Fr& f) {
        u64 c[4];
        f.to_canonical(c);
        ab(c, 32);
    }
    void append_g1(const G1Aff& p, bool infinity = false) {
        if (infinity) {
            uint8_t z[64] = {0};
            ab(z, 64);
        } else {
            u64 c[4];
            p.x.to_canonical(c);
            ab(c, 32);
            p.y.to_canonical(c);
            ab(c, 32);
        }
    }
    Fr challenge() {
        Keccak k;
        k.update(state, 32);
        if (!buf.empty()) k.update(buf.data(), buf.size());
        uint8_t h[32];
        k.digest(h);
        memcpy(state, h, 32);
        buf.clear();
        return fr_from_hash(h);
    }
};

struct HostDrbg {
    uint64_t seed;
    uint32_t ctr = 0;
    explicit HostDrbg(uint64_t s) : seed(s) {}
    Fr next() {
        uint8_t msg[21];
        memcpy(msg, "rng-blind", 9);
        memcpy(msg + 9, &seed, 8);
        memcpy(msg + 17, &ctr, 4);
        ctr++;
        uint8_t h[32];
        keccak256_h(msg, 21, h);
        return fr_from_hash(h);
    }
};

// ---- host polynomial helpers ----
inline Fr hpoly_eval(const std::vector<Fr>& p, const Fr& x) {
    Fr acc = Fr::zero();
    for (size_t i = p.size(); i-- > 0;) acc = acc.mul(x).add(p[i]);
    return acc;
}
inline std::vector<Fr> hpoly_div_linear(const std::vector<Fr>& p, const Fr& zeta) {
    if (p.size() < 2) return {};
    size_t d = p.size() - 1;
    std::vector<Fr> q(d, Fr::zero());
    q[d - 1] = p[d];
    for (size_t i = d - 1; i >= 1; --i) q[i - 1] = p[i].add(zeta.mul(q[i]));
    return q;
}
inline void hpoly_add_scaled(std::vector<Fr>& acc, const std::vector<Fr>& p, const Fr& s) {
    if (acc.size() < p.size()) acc.resize(p.size(), Fr::zero());
    for (size_t i = 0; i < p.size(); ++i) acc[i] = acc[i].add(p[i].mul(s));
}
inline std::vector<Fr> hbatch_inverse(std::vector<Fr> v) {
    size_t m = v.size();
    std::vector<Fr> pre(m);
    Fr acc = Fr::one();
    for (size_t i = 0; i < m; ++i) {
        pre[i] = acc;
        acc = acc.mul(v[i]);
    }
    Fr inv = acc.inverse();
    for (size_t i = m; i-- > 0;) {
        Fr orig = v[i];
        v[i] = inv.mul(pre[i]);
        inv = inv.mul(orig);
    }
    return v;
}

}  // namespace rng
