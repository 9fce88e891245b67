// CPU ORACLE — test infrastructure only (see field.hpp header).
//
// Restates: arkworks 0.4.2 ark-poly Radix2EvaluationDomain fft/ifft over
// BN254 Fr.  Convention pinned here (and mirrored by
// the HIP path): natural-order coefficients in -> natural-order evaluations
// out, out[k] = sum_j in[j] * w^(jk) with w = FR_TWO_ADIC_ROOT^(2^28 / n);
// inverse uses w^-1 and scales by n^-1.
#pragma once
#include <vector>
#include <cassert>
#include "field.hpp"

namespace oracle {

inline Fr fr_root_of_unity(size_t n) {
    // n must be a power of two <= 2^28
    static constexpr u64 root_c[4] = FR_TWO_ADIC_ROOT;
    Fr root = Fr::from_canonical(root_c);
    size_t log_n = 0;
    while ((size_t(1) << log_n) < n) log_n++;
    assert((size_t(1) << log_n) == n && log_n <= FR_TWO_ADICITY);
    for (size_t i = log_n; i < FR_TWO_ADICITY; ++i) root = root.square();
    return root;
}

inline void bit_reverse_permute(Fr* a, size_t n) {
    size_t log_n = 0;
    while ((size_t(1) << log_n) < n) log_n++;
    for (size_t i = 0; i < n; ++i) {
        size_t j = 0;
        for (size_t b = 0; b < log_n; ++b) j |= ((i >> b) & 1) << (log_n - 1 - b);
        if (j > i) std::swap(a[i], a[j]);
    }
}

// In-place DIT radix-2 NTT; `w` is a primitive n-th root (Montgomery form).
inline void ntt_core(Fr* a, size_t n, const Fr& w) {
    bit_reverse_permute(a, n);
    for (size_t len = 2; len <= n; len <<= 1) {
        Fr wl = w;
        for (size_t i = len; i < n; i <<= 1) wl = wl.square();  // w^(n/len)
        for (size_t start = 0; start < n; start += len) {
            Fr t = Fr::one();
            for (size_t k = 0; k < len / 2; ++k) {
                Fr u = a[start + k];
                Fr v = a[start + k + len / 2] * t;
                a[start + k] = u + v;
                a[start + k + len / 2] = u - v;
                t = t * wl;
            }
        }
    }
}

inline void ntt_forward(Fr* a, size_t n) { ntt_core(a, n, fr_root_of_unity(n)); }

inline void ntt_inverse(Fr* a, size_t n) {
    Fr w = fr_root_of_unity(n);
    ntt_core(a, n, w.inverse());
    Fr ninv = Fr::from_u64((u64)n).inverse();
    for (size_t i = 0; i < n; ++i) a[i] = a[i] * ninv;
}

// Coset NTT: evaluate on {g * w^k} — multiply a[j] by g^j first.
inline void coset_ntt_forward(Fr* a, size_t n, const Fr& g) {
    Fr gj = Fr::one();
    for (size_t i = 0; i < n; ++i) {
        a[i] = a[i] * gj;
        gj = gj * g;
    }
    ntt_forward(a, n);
}

inline void coset_ntt_inverse(Fr* a, size_t n, const Fr& g) {
    ntt_inverse(a, n);
    Fr ginv = g.inverse();
    Fr gj = Fr::one();
    for (size_t i = 0; i < n; ++i) {
        a[i] = a[i] * gj;
        gj = gj * ginv;
    }
}

}  // namespace oracle
