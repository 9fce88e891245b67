// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Keccak-256 (the pre-NIST padding variant used by Ethereum/Solidity), as
// required by the reference's Fiat-Shamir transcript
// (`SolidityTranscript`, consumed at
//  crates/circuits/circuit-types/src/traits.rs:996,1012).
// Validated in tests/ against published keccak256 known-answer vectors.
#pragma once
#include <cstdint>
#include <cstring>
#include <cstddef>

namespace oracle {

struct Keccak256 {
    static constexpr int RATE = 136;  // 1088-bit rate
    uint64_t st[25];
    uint8_t buf[RATE];
    size_t buflen;

    Keccak256() { reset(); }
    void reset() {
        memset(st, 0, sizeof(st));
        buflen = 0;
    }

    // n == 0 must not shift by 64 (UB; clang -O3 miscompiles it)
    static inline uint64_t rotl(uint64_t x, int n) {
        return n == 0 ? x : (x << n) | (x >> (64 - n));
    }

    void permute() {
        static const uint64_t RC[24] = {
            0x0000000000000001ULL, 0x0000000000008082ULL, 0x800000000000808aULL,
            0x8000000080008000ULL, 0x000000000000808bULL, 0x0000000080000001ULL,
            0x8000000080008081ULL, 0x8000000000008009ULL, 0x000000000000008aULL,
            0x0000000000000088ULL, 0x0000000080008009ULL, 0x000000008000000aULL,
            0x000000008000808bULL, 0x800000000000008bULL, 0x8000000000008089ULL,
            0x8000000000008003ULL, 0x8000000000008002ULL, 0x8000000000000080ULL,
            0x000000000000800aULL, 0x800000008000000aULL, 0x8000000080008081ULL,
            0x8000000000008080ULL, 0x0000000080000001ULL, 0x8000000080008008ULL};
        static const int r[5][5] = {{0, 36, 3, 41, 18},
                                    {1, 44, 10, 45, 2},
                                    {62, 6, 43, 15, 61},
                                    {28, 55, 25, 21, 56},
                                    {27, 20, 39, 8, 14}};
        for (int round = 0; round < 24; ++round) {
            uint64_t C[5], D[5];
            for (int x = 0; x < 5; ++x)
                C[x] = st[x] ^ st[x + 5] ^ st[x + 10] ^ st[x + 15] ^ st[x + 20];
            for (int x = 0; x < 5; ++x) D[x] = C[(x + 4) % 5] ^ rotl(C[(x + 1) % 5], 1);
            for (int x = 0; x < 5; ++x)
                for (int y = 0; y < 5; ++y) st[x + 5 * y] ^= D[x];
            // rho + pi
            uint64_t B[25];
            for (int x = 0; x < 5; ++x)
                for (int y = 0; y < 5; ++y)
                    B[y + 5 * ((2 * x + 3 * y) % 5)] = rotl(st[x + 5 * y], r[x][y]);
            // chi
            for (int x = 0; x < 5; ++x)
                for (int y = 0; y < 5; ++y)
                    st[x + 5 * y] = B[x + 5 * y] ^ ((~B[(x + 1) % 5 + 5 * y]) & B[(x + 2) % 5 + 5 * y]);
            st[0] ^= RC[round];
        }
    }

    void absorb_block(const uint8_t* p) {
        for (int i = 0; i < RATE / 8; ++i) {
            uint64_t v;
            memcpy(&v, p + 8 * i, 8);
            st[i] ^= v;  // little-endian host assumed (x86/amdgcn)
        }
        permute();
    }

    void update(const uint8_t* data, size_t len) {
        while (len > 0) {
            size_t take = RATE - buflen;
            if (take > len) take = len;
            memcpy(buf + buflen, data, take);
            buflen += take;
            data += take;
            len -= take;
            if (buflen == RATE) {
                absorb_block(buf);
                buflen = 0;
            }
        }
    }

    void finalize(uint8_t out[32]) {
        // keccak padding: 0x01 ... 0x80 (multi-rate pad with domain bit 01)
        uint8_t block[RATE];
        memcpy(block, buf, buflen);
        memset(block + buflen, 0, RATE - buflen);
        block[buflen] ^= 0x01;
        block[RATE - 1] ^= 0x80;
        absorb_block(block);
        memcpy(out, st, 32);
    }
};

inline void keccak256(const uint8_t* data, size_t len, uint8_t out[32]) {
    Keccak256 k;
    k.update(data, len);
    k.finalize(out);
}

}  // namespace oracle
