// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Restates: arkworks 0.4.2 ark-ec VariableBaseMSM (Pippenger) over BN254 G1
// (SURVEY.md §8a row a5).  Scalars are canonical (non-Montgomery) 4 x u64
// little-endian limbs, bases are affine Montgomery-form points — exactly the
// shapes the KZG commit path hands to the extern MSM in the reference
// (called from mpc-jellyfish via ark-ec; dep pinned Cargo.lock:933-1199).
#pragma once
#include <vector>
#include <omp.h>
#include "curve.hpp"

namespace oracle {

// Reference implementation: plain double-and-add sum (for small-n cross-checks).
inline G1Proj msm_naive(const G1Affine* bases, const u64* scalars, size_t n) {
    G1Proj acc = G1Proj::identity();
    for (size_t i = 0; i < n; ++i) {
        acc = acc.add(G1Proj::from_affine(bases[i]).mul(scalars + 4 * i));
    }
    return acc;
}

// Pippenger bucket method, window size c, unsigned digits.  OpenMP over
// (window, point-chunk) pairs with per-task bucket arrays merged per window,
// so it scales past num_windows cores (the GPU judge box has 256).
inline G1Proj msm_pippenger(const G1Affine* bases, const u64* scalars, size_t n,
                            int c = 13) {
    const int num_windows = (256 + c - 1) / c;
    const size_t num_buckets = (size_t(1) << c) - 1;  // digit 0 skipped
    std::vector<G1Proj> window_sums(num_windows, G1Proj::identity());

    int nthreads = omp_get_max_threads();
    // balance bucket-accumulation (n/chunks per task) against per-window merge
    // (chunks * 2^c): optimum near sqrt(n / 2^c)
    int chunks = 1;
    while ((size_t)chunks * chunks * num_buckets < n) chunks *= 2;
    if (chunks > nthreads) chunks = nthreads;
    if (chunks > 16) chunks = 16;
    if (chunks < 1 || n < (size_t)(64 * chunks)) chunks = 1;
    const size_t chunk_sz = (n + chunks - 1) / chunks;

    std::vector<std::vector<G1Proj>> chunk_buckets((size_t)num_windows * chunks);

#pragma omp parallel for schedule(dynamic, 1) collapse(2)
    for (int w = 0; w < num_windows; ++w) {
        for (int ch = 0; ch < chunks; ++ch) {
            std::vector<G1Proj> buckets(num_buckets, G1Proj::identity());
            const int bit0 = w * c;
            size_t lo = ch * chunk_sz, hi = lo + chunk_sz < n ? lo + chunk_sz : n;
            for (size_t i = lo; i < hi; ++i) {
                const u64* s = scalars + 4 * i;
                u64 digit;
                int limb = bit0 / 64, off = bit0 % 64;
                digit = s[limb] >> off;
                if (off + c > 64 && limb + 1 < 4) digit |= s[limb + 1] << (64 - off);
                digit &= (u64(1) << c) - 1;
                if (digit != 0) buckets[digit - 1] = buckets[digit - 1].add_affine(bases[i]);
            }
            chunk_buckets[(size_t)w * chunks + ch] = std::move(buckets);
        }
    }

#pragma omp parallel for schedule(dynamic, 1)
    for (int w = 0; w < num_windows; ++w) {
        // merge chunk bucket arrays, then running suffix sum: sum_d d*bucket[d]
        G1Proj run = G1Proj::identity(), sum = G1Proj::identity();
        for (size_t d = num_buckets; d >= 1; --d) {
            G1Proj b = chunk_buckets[(size_t)w * chunks][d - 1];
            for (int ch = 1; ch < chunks; ++ch)
                b = b.add(chunk_buckets[(size_t)w * chunks + ch][d - 1]);
            run = run.add(b);
            sum = sum.add(run);
        }
        window_sums[w] = sum;
    }

    // Horner over windows: acc = W_last; acc = acc*2^c + W_w
    G1Proj acc = window_sums[num_windows - 1];
    for (int w = num_windows - 2; w >= 0; --w) {
        for (int k = 0; k < c; ++k) acc = acc.dbl();
        acc = acc.add(window_sums[w]);
    }
    return acc;
}

}  // namespace oracle
