// PRODUCT PATH — TurboPlonk quotient evaluation on gfx950.
//
// Pointwise evaluation of the quotient numerator over the 8n coset domain
// (the dominant non-NTT/MSM column work of the prover):
//   num(t) = gate(t) + alpha [ z f - z_shift g ] + alpha^2 (z-1) L1
//   quot(t) = num(t) * zh_inv[t % 8]
// One thread per coset point; all operands are coset-evaluation arrays laid
// out contiguously (sel: 13*m, sig: 5*m, w: 5*m, z/pi/l1: m, xpow: m = the
// coset points g*w_m^t for the beta*k_j*X term).
#include <hip/hip_runtime.h>
#include "gpu_field.hpp"

namespace rng {

struct QuotChal {  // challenge pack (kernel arg)
    Fr beta, gamma, alpha, alpha2;
    Fr k[5];
    Fr zh_inv[8];
};

__global__ __launch_bounds__(256) void k_quotient(
    const Fr* sel, const Fr* sig, const Fr* w, const Fr* z, const Fr* pi,
    const Fr* l1, const Fr* xpow, Fr* out, uint32_t m, QuotChal ch) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= m) return;
    auto p5 = [](const Fr& v) {
        Fr v2 = v.sqr();
        return v2.sqr().mul(v);
    };
    Fr w0 = w[t], w1 = w[m + t], w2 = w[2 * m + t], w3 = w[3 * m + t],
       w4 = w[4 * m + t];
    // gate equation: selector order = plonk_circuit.hpp enum Sel
    Fr gate = sel[11 * m + t].add(pi[t]);
    gate = gate.add(sel[0 * m + t].mul(w0)).add(sel[1 * m + t].mul(w1));
    gate = gate.add(sel[2 * m + t].mul(w2)).add(sel[3 * m + t].mul(w3));
    gate = gate.add(sel[4 * m + t].mul(w0.mul(w1))).add(sel[5 * m + t].mul(w2.mul(w3)));
    gate = gate.add(sel[6 * m + t].mul(p5(w0))).add(sel[7 * m + t].mul(p5(w1)));
    gate = gate.add(sel[8 * m + t].mul(p5(w2))).add(sel[9 * m + t].mul(p5(w3)));
    gate = gate.add(sel[12 * m + t].mul(w0.mul(w1).mul(w2).mul(w3).mul(w4)));
    gate = gate.sub(sel[10 * m + t].mul(w4));
    // permutation
    Fr x = xpow[t];
    Fr f = Fr::one(), g = Fr::one();
    const Fr* ws[5] = {&w0, &w1, &w2, &w3, &w4};
    for (int j = 0; j < 5; ++j) {
        f = f.mul(ws[j]->add(ch.beta.mul(ch.k[j]).mul(x)).add(ch.gamma));
        g = g.mul(ws[j]->add(ch.beta.mul(sig[j * m + t])).add(ch.gamma));
    }
    Fr zshift = z[(t + 8) % m];
    Fr perm = ch.alpha.mul(z[t].mul(f).sub(zshift.mul(g)));
    Fr l1term = ch.alpha2.mul(z[t].sub(Fr::one())).mul(l1[t]);
    out[t] = gate.add(perm).add(l1term).mul(ch.zh_inv[t & 7]);
}

// cohort-batched quotient: K proofs in one launch (K*m threads) — a single
// proof's m=8n grid is only ~0.5 waves/SIMD at n=4096, pure latency; the
// cohort profile showed 483 small k_quotient launches at 14% of GPU time.
// Wire/z/PI cosets are laid out per proof at stride 7m (slots 0-4 wires,
// 5 z, 6 PI — plonk_prove_cohort_impl); challenges per proof from HBM.
__global__ __launch_bounds__(256) void k_quotient_batch(
    const Fr* sel, const Fr* sig, const Fr* coset_all, const Fr* l1,
    const Fr* xpow, Fr* out_all, uint32_t m, uint32_t K, const QuotChal* chs) {
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (tid >= (uint64_t)K * m) return;
    uint32_t p = (uint32_t)(tid / m);
    uint32_t t = (uint32_t)(tid - (uint64_t)p * m);
    const QuotChal& ch = chs[p];
    const Fr* w = coset_all + 7 * (uint64_t)p * m;
    const Fr* z = w + 5 * (uint64_t)m;
    const Fr* pi = w + 6 * (uint64_t)m;
    Fr* out = out_all + (uint64_t)p * m;
    auto p5 = [](const Fr& v) {
        Fr v2 = v.sqr();
        return v2.sqr().mul(v);
    };
    Fr w0 = w[t], w1 = w[m + t], w2 = w[2 * m + t], w3 = w[3 * m + t],
       w4 = w[4 * m + t];
    Fr gate = sel[11 * m + t].add(pi[t]);
    gate = gate.add(sel[0 * m + t].mul(w0)).add(sel[1 * m + t].mul(w1));
    gate = gate.add(sel[2 * m + t].mul(w2)).add(sel[3 * m + t].mul(w3));
    gate = gate.add(sel[4 * m + t].mul(w0.mul(w1))).add(sel[5 * m + t].mul(w2.mul(w3)));
    gate = gate.add(sel[6 * m + t].mul(p5(w0))).add(sel[7 * m + t].mul(p5(w1)));
    gate = gate.add(sel[8 * m + t].mul(p5(w2))).add(sel[9 * m + t].mul(p5(w3)));
    gate = gate.add(sel[12 * m + t].mul(w0.mul(w1).mul(w2).mul(w3).mul(w4)));
    gate = gate.sub(sel[10 * m + t].mul(w4));
    Fr x = xpow[t];
    Fr f = Fr::one(), g = Fr::one();
    const Fr* ws[5] = {&w0, &w1, &w2, &w3, &w4};
    for (int j = 0; j < 5; ++j) {
        f = f.mul(ws[j]->add(ch.beta.mul(ch.k[j]).mul(x)).add(ch.gamma));
        g = g.mul(ws[j]->add(ch.beta.mul(sig[j * m + t])).add(ch.gamma));
    }
    Fr zshift = z[(t + 8) % m];
    Fr perm = ch.alpha.mul(z[t].mul(f).sub(zshift.mul(g)));
    Fr l1term = ch.alpha2.mul(z[t].sub(Fr::one())).mul(l1[t]);
    out[t] = gate.add(perm).add(l1term).mul(ch.zh_inv[t & 7]);
}

// elementwise multiply: data[i] *= table[i] (coset scaling)
__global__ __launch_bounds__(256) void k_mul_pointwise(Fr* data, const Fr* table,
                                                       uint32_t count) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < count) data[i] = data[i].mul(table[i]);
}

// multiply every element by a constant
__global__ __launch_bounds__(256) void k_scale_const(Fr* data, Fr c, uint32_t count) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < count) data[i] = data[i].mul(c);
}

// zero-extend helper: copy src (len) into dst (m), zero the rest
__global__ __launch_bounds__(256) void k_copy_pad(const Fr* src, uint32_t len, Fr* dst,
                                                  uint32_t m) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= m) return;
    dst[i] = (i < len) ? src[i] : Fr::zero();
}

// cohort R1: blind + pad 5k wire polynomials on device.  in: coeffs
// (5k polys x n contiguous), blinders (2 per poly); out: 5k polys at
// stride n+2 with wp[0]-=b0, wp[1]-=b1, wp[n]+=b0, wp[n+1]+=b1
// (traits.rs:994 blinding shape; DRBG order preserved host-side).
// Writes TWO copies: `out` (contiguous stride n+2, the commit staging) and
// `coset_out` slot (p*7 + wire) at stride n+3 zero-padded — the R3 coset
// batch reads wires/z/PI from that layout with no host round trip.
__global__ __launch_bounds__(256) void k_blind_wires_batch(
    const Fr* coeffs, const Fr* blinders, Fr* out, Fr* coset_out, uint32_t n,
    uint64_t total /* 5k*(n+3) */) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= total) return;
    uint64_t stride = (uint64_t)n + 3;
    uint64_t g = i / stride;
    uint32_t j = (uint32_t)(i - g * stride);
    Fr v = (j < n) ? coeffs[g * n + j] : Fr::zero();
    Fr b0 = blinders[2 * g], b1 = blinders[2 * g + 1];
    if (j == 0) v = v.sub(b0);
    if (j == 1) v = v.sub(b1);
    if (j == n) v = v.add(b0);
    if (j == n + 1) v = v.add(b1);
    uint64_t p = g / 5;
    coset_out[(p * 7 + (g - p * 5)) * stride + j] = v;
    if (j < n + 2) out[g * (uint64_t)(n + 2) + j] = v;
}

// cohort R2: blind the k grand-product polynomials on device; same double
// write (commit staging at stride n+3, coset slot p*7+5).  Blinder order
// mirrors plonk_prove_impl: draws (b2,b3,b4) apply as zp[0]-=b4, zp[1]-=b3,
// zp[2]-=b2, zp[n]+=b4, zp[n+1]+=b3, zp[n+2]+=b2.
__global__ __launch_bounds__(256) void k_blind_z_batch(
    const Fr* coeffs, const Fr* blinders /* 3 per proof: b2,b3,b4 */, Fr* out,
    Fr* coset_out, uint32_t n, uint64_t total /* k*(n+3) */) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= total) return;
    uint64_t stride = (uint64_t)n + 3;
    uint64_t p = i / stride;
    uint32_t j = (uint32_t)(i - p * stride);
    Fr v = (j < n) ? coeffs[p * n + j] : Fr::zero();
    Fr b2 = blinders[3 * p], b3 = blinders[3 * p + 1], b4 = blinders[3 * p + 2];
    if (j == 0) v = v.sub(b4);
    if (j == 1) v = v.sub(b3);
    if (j == 2) v = v.sub(b2);
    if (j == n) v = v.add(b4);
    if (j == n + 1) v = v.add(b3);
    if (j == n + 2) v = v.add(b2);
    out[i] = v;
    coset_out[(p * 7 + 5) * stride + j] = v;
}

// cohort R2: stage the k public-input polynomials (n coeffs each) into
// their coset slots (p*7+6) zero-padded to stride n+3
__global__ __launch_bounds__(256) void k_pi_to_coset(const Fr* coeffs,
                                                     Fr* coset_out, uint32_t n,
                                                     uint64_t total /* k*(n+3) */) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= total) return;
    uint64_t stride = (uint64_t)n + 3;
    uint64_t p = i / stride;
    uint32_t j = (uint32_t)(i - p * stride);
    coset_out[(p * 7 + 6) * stride + j] = (j < n) ? coeffs[p * n + j] : Fr::zero();
}

// cohort R3: split each proof's quotient (m coeffs) into 5 chunks of
// degree <= n+2 with the linking blinders applied on device: chunk i gets
// qc[0] -= b_{i-1} and (i<4) qc[n+2] += b_i.  out: 5k polys at stride n+3
// (chunk 4's top coefficient is zero).
__global__ __launch_bounds__(256) void k_quot_chunks_batch(
    const Fr* quot, const Fr* blinders /* 4 per proof */, Fr* out, uint32_t n,
    uint32_t m, uint64_t total /* 5k*(n+3) */) {
    uint64_t idx = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    uint64_t stride = (uint64_t)n + 3;
    uint64_t g = idx / stride;           // p*5 + i
    uint32_t j = (uint32_t)(idx - g * stride);
    uint64_t p = g / 5;
    uint32_t i = (uint32_t)(g - p * 5);
    Fr v = Fr::zero();
    if (j < n + 2) v = quot[p * m + (uint64_t)i * (n + 2) + j];
    if (j == 0 && i > 0) v = v.sub(blinders[4 * p + i - 1]);
    if (j == n + 2 && i < 4) v = v.add(blinders[4 * p + i]);
    out[idx] = v;
}

// batched coset scaling for nb polynomials of `stride` coefficients each
// stored back to back: data[b*stride + j] *= table[j]  (table >= stride long)
__global__ __launch_bounds__(256) void k_mul_pointwise_mod(Fr* data, const Fr* table,
                                                           uint32_t stride,
                                                           uint64_t total) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < total) data[i] = data[i].mul(table[i % stride]);
}

// batched zero-extension: nb polys of `len` coefficients at stride `len`
// in src -> nb polys padded to m in dst
__global__ __launch_bounds__(256) void k_copy_pad_batch(const Fr* src, uint32_t len,
                                                        Fr* dst, uint32_t m,
                                                        uint64_t total /* nb*m */) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= total) return;
    uint64_t b = i / m;
    uint32_t j = (uint32_t)(i - b * m);
    dst[i] = (j < len) ? src[b * len + j] : Fr::zero();
}

// ---- field-mul microbenchmark kernels (A/B the Montgomery formulations) ----
// Each thread runs `iters` muls: dep = a dependent chain (latency),
// otherwise 4 independent chains (throughput).
template <int VARIANT>
__global__ __launch_bounds__(256) void k_bench_frmul(Fr* io, uint32_t iters, int dep) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    Fr a = io[t & 1023];
    Fr b = io[(t + 1) & 1023];
    if (dep) {
        for (uint32_t i = 0; i < iters; ++i)
            a = (VARIANT == 0) ? a.mul_cios(b) : (VARIANT == 1 ? a.mul32(b) : a.mul_sos(b));
    } else {
        Fr c = io[(t + 2) & 1023], d = io[(t + 3) & 1023];
        Fr e = io[(t + 4) & 1023], f = io[(t + 5) & 1023];
        for (uint32_t i = 0; i < iters / 4; ++i) {
            if (VARIANT == 0) {
                a = a.mul_cios(b);
                c = c.mul_cios(b);
                d = d.mul_cios(b);
                e = e.mul_cios(f);
            } else if (VARIANT == 1) {
                a = a.mul32(b);
                c = c.mul32(b);
                d = d.mul32(b);
                e = e.mul32(f);
            } else {
                a = a.mul_sos(b);
                c = c.mul_sos(b);
                d = d.mul_sos(b);
                e = e.mul_sos(f);
            }
        }
        a = a.add(c).add(d).add(e);
    }
    io[t & 1023] = a;  // keep alive
}

}  // namespace rng
