// PRODUCT PATH — MI355X-native Pippenger MSM over BN254 G1.
//
// Replaces: arkworks ark-ec VariableBaseMSM as consumed through the
// reference's KZG commitments (BASELINE config #2).
//
// Structure (sort-based, designed for gfx950; B polynomials batch through
// one pipeline via key group g = poly*W + window):
//  1. k_msm_digits: signed windowed digit decomposition (c in {8,13,16} so
//     the TOP window stays populated — see msm_auto_c), one
//     (key = group<<16 | magnitude, val = sign<<31 | point index) pair per
//     nonzero digit; zero digits get the dynamic sentinel (G<<16).
//  2. rocPRIM device radix sort over the live key bits only (3 passes) —
//     groups every bucket's points together; no atomics or EC critical
//     sections anywhere.
//  3. segment heads via flag+select; long segments split into capped
//     sub-segments (k_msm_seg_lengths / k_msm_make_subs), reduced
//     longest-first one lane each with Jacobian mixed adds
//     (k_msm_bucket_reduce), then merged per bucket longest-first
//     (k_msm_seg_merge).
//  4. k_msm_window_chunks: per (window, chunk) suffix running sums ->
//     (sum, weighted-sum) partials; chunk size adapts to keep >=32k lanes.
//  5. k_msm_window_combine: per-window fold of the chunk partials
//     (sub-block count adapts); the final Horner across windows runs on
//     the HOST (a 1-lane dependent EC chain is far faster on a host core).
// A GLV-endomorphism variant and a fused suffix-scan fold exist behind
// RNG_MSM_GLV / RNG_MSM_FOLD (both measured slower on MI355X).
#include <hip/hip_runtime.h>
#include <rocprim/device/device_radix_sort.hpp>
#include <rocprim/device/device_select.hpp>
#include <rocprim/device/device_scan.hpp>
#include <rocprim/iterator/counting_iterator.hpp>
#include "gpu_curve.hpp"

namespace rng {

// zero digits get the sentinel key (G << 16) — one past the last real
// group, so it sorts after every real (group<<16|mag) key while keeping the
// radix sort to 16 + ceil(log2(G+1)) bits (3 passes instead of 4 for every
// workload here)
__host__ __device__ inline uint32_t msm_sentinel(uint32_t G) { return G << 16; }
constexpr uint32_t MSM_CHUNK = 16;            // buckets per window-sum thread

// window size by problem size.  Besides balancing bucket-phase work
// (~ n*W(c)) against aggregation (~ W(c)*2^(c-1)), c must keep the TOP
// window well-populated: with 254-bit scalars the top window holds only
// 254-(W-1)*c significant bits, and when that is small every scalar's top
// digit lands in a handful of buckets — giant segments whose serial merge
// chains dominate (measured 3x slowdown at c=9/11 vs c=8 for n=4096).
// Top-window distinct digits D(c) = 2^(254-(W-1)*c): c=8 -> 64, c=13 -> 128,
// c=16 -> 2^14; those three tiers are the sweet spots.
__host__ __device__ inline uint32_t msm_auto_c(uint64_t n) {
    if (n <= (1ull << 16)) return 8;
    if (n <= (1ull << 18)) return 13;
    return 16;
}

// ---- 1. digit decomposition ----
// scalars: canonical LE 4xu64. keys/vals: n*W entries, window-major
// (out[w*n + i]) so writes coalesce per window.
// B polys of n scalars each share one base array; digits of poly b window w
// go to key group g = b*W + w so one sort/reduce handles the whole batch.
__global__ __launch_bounds__(256) void k_msm_digits(const uint64_t* scalars, uint32_t n, uint32_t c,
                             uint32_t W, uint32_t B, uint32_t* keys, uint32_t* vals) {
    uint32_t idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n * B) return;
    uint32_t b = idx / n, i = idx % n;
    uint64_t s[4];
    s[0] = scalars[4 * idx];
    s[1] = scalars[4 * idx + 1];
    s[2] = scalars[4 * idx + 2];
    s[3] = scalars[4 * idx + 3];
    uint32_t carry = 0;
    uint32_t half = 1u << (c - 1);
    uint64_t cmask = (c == 64) ? ~0ull : ((1ull << c) - 1);
    for (uint32_t w = 0; w < W; ++w) {
        uint32_t bit0 = w * c;
        uint32_t limb = bit0 >> 6, off = bit0 & 63;
        uint64_t raw = s[limb] >> off;
        if (off + c > 64 && limb + 1 < 4) raw |= s[limb + 1] << (64 - off);
        raw = (raw & cmask) + carry;
        uint32_t mag, sign;
        if (raw >= half) {  // treat as negative digit raw - 2^c unless raw == half
            if (raw > half) {
                mag = (uint32_t)((1ull << c) - raw);
                sign = 1;
                carry = 1;
            } else {  // raw == half: use +half, no carry (mag fits 16 bits for c<=16)
                mag = half;
                sign = 0;
                carry = 0;
            }
        } else {
            mag = (uint32_t)raw;
            sign = 0;
            carry = 0;
        }
        uint64_t o = ((uint64_t)b * W + w) * n + i;
        keys[o] = mag == 0 ? msm_sentinel(B * W) : (((b * W + w) << 16) | mag);
        vals[o] = (sign << 31) | i;
    }
    // carry out of the top window must be zero for scalars < 2^(W*c-1)
}

// ---- 1-alt. binned (counting-scatter) pipeline ----
// Replaces the 3-pass radix sort with histogram + exclusive scan + scatter:
// ~3x less traffic per entry, zero digits skipped outright (no sentinel),
// and segment heads/lengths fall out of the histogram for free.  Intra-
// bucket order becomes nondeterministic (atomic cursors), which is fine:
// EC addition is exactly associative/commutative, and every consumer of a
// bucket/window sum normalizes to a unique affine record.

// digits + histogram over dense bucket ids bid = group*2^(c-1) + (mag-1)
__global__ __launch_bounds__(256) void k_msm_digits_hist(
    const uint64_t* scalars, uint32_t n, uint32_t c, uint32_t W, uint32_t B,
    uint32_t* bids, uint32_t* vals, uint32_t* counts) {
    uint32_t idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n * B) return;
    uint32_t b = idx / n, i = idx % n;
    uint64_t s[4];
    s[0] = scalars[4 * idx];
    s[1] = scalars[4 * idx + 1];
    s[2] = scalars[4 * idx + 2];
    s[3] = scalars[4 * idx + 3];
    uint32_t carry = 0;
    uint32_t half = 1u << (c - 1);
    uint64_t cmask = (c == 64) ? ~0ull : ((1ull << c) - 1);
    for (uint32_t w = 0; w < W; ++w) {
        uint32_t bit0 = w * c;
        uint32_t limb = bit0 >> 6, off = bit0 & 63;
        uint64_t raw = s[limb] >> off;
        if (off + c > 64 && limb + 1 < 4) raw |= s[limb + 1] << (64 - off);
        raw = (raw & cmask) + carry;
        uint32_t mag, sign;
        if (raw >= half) {
            if (raw > half) {
                mag = (uint32_t)((1ull << c) - raw);
                sign = 1;
                carry = 1;
            } else {
                mag = half;
                sign = 0;
                carry = 0;
            }
        } else {
            mag = (uint32_t)raw;
            sign = 0;
            carry = 0;
        }
        uint64_t o = ((uint64_t)b * W + w) * n + i;
        if (mag == 0) {
            bids[o] = 0xFFFFFFFFu;  // skip marker
        } else {
            uint32_t bid = (b * W + w) * half + (mag - 1);
            bids[o] = bid;
            atomicAdd(&counts[bid], 1u);
        }
        vals[o] = (sign << 31) | i;
    }
}

// scatter entries to their bucket segment (cursor = copy of scanned offsets)
__global__ __launch_bounds__(256) void k_msm_scatter(const uint32_t* bids,
                                                     const uint32_t* vals,
                                                     uint64_t total,
                                                     uint32_t* cursor,
                                                     uint32_t* vals_out) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= total) return;
    uint32_t bid = bids[t];
    if (bid == 0xFFFFFFFFu) return;
    uint32_t pos = atomicAdd(&cursor[bid], 1u);
    vals_out[pos] = vals[t];
}

__global__ void k_nonzero_flags(const uint32_t* counts, uint32_t nb, uint8_t* flags) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t < nb) flags[t] = counts[t] != 0 ? 1 : 0;
}

// heads/lens/nsub for the compacted non-empty buckets
__global__ void k_msm_seg_from_counts(const uint32_t* bucket_ids,
                                      const uint32_t* offsets,
                                      const uint32_t* counts,
                                      const uint32_t* head_count, uint32_t* heads,
                                      uint32_t* lens, uint32_t* nsub,
                                      uint32_t seg_cap) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= *head_count) return;
    uint32_t bid = bucket_ids[t];
    uint32_t len = counts[bid];
    heads[t] = offsets[bid];
    lens[t] = len;
    nsub[t] = (len + seg_cap - 1) / seg_cap;
}

// merge sub-partials per segment -> bucket, addressed by dense bucket id
__global__ __launch_bounds__(256) void k_msm_seg_merge2(
    const uint32_t* bucket_ids, const uint32_t* sub_off, const uint32_t* nsub,
    const uint32_t* head_order /* by nsub desc */, uint32_t head_count,
    const G1Jac* partials2, G1Jac* buckets) {
    uint32_t tt = blockIdx.x * blockDim.x + threadIdx.x;
    if (tt >= head_count) return;
    uint32_t t = head_order[tt];
    uint32_t off = sub_off[t], ns = nsub[t];
    G1Jac acc = partials2[off];
    for (uint32_t k = 1; k < ns; ++k) acc = acc.add(partials2[off + k]);
    buckets[bucket_ids[t]] = acc;
}

// ---- 1b. GLV endomorphism path (include/bn254_glv.h, self-verified) ----
// k = k1 + lambda*k2 (mod r), |ki| < 2^127; k*P = k1*P + k2*phi(P) with
// phi(x,y) = (beta*x, y).  Halves the Pippenger windows (127-bit halves),
// so the aggregation stages (buckets, window sums, Horner) halve too.
#include "../../include/bn254_glv.h"

#if GLV_A1_NEG || !GLV_B1_NEG || GLV_A2_NEG || GLV_B2_NEG
#error "GLV kernels assume sign pattern a1>=0, b1<=0, a2>=0, b2>=0"
#endif

// (g * k + 2^(SHIFT-1)) >> SHIFT for 4-limb g,k; result < 2^127 (2 limbs)
__host__ __device__ inline void glv_mul_shift(const u64 g[4], const u64 k[4], u64 out[2]) {
    u64 acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int i = 0; i < 4; ++i) {
        unsigned __int128 carry = 0;
        for (int j = 0; j < 4; ++j) {
            unsigned __int128 t = (unsigned __int128)g[i] * k[j] + acc[i + j] + (u64)carry;
            acc[i + j] = (u64)t;
            carry = t >> 64;
        }
        int idx = i + 4;
        while (carry) {
            unsigned __int128 t = (unsigned __int128)acc[idx] + (u64)carry;
            acc[idx] = (u64)t;
            carry = t >> 64;
            ++idx;
        }
    }
    // + 2^319 (rounding), then >> 320
    unsigned __int128 t = (unsigned __int128)acc[4] + (1ull << 63);
    acc[4] = (u64)t;
    u64 cy = (u64)(t >> 64);
    for (int idx = 5; cy && idx < 8; ++idx) {
        t = (unsigned __int128)acc[idx] + cy;
        acc[idx] = (u64)t;
        cy = (u64)(t >> 64);
    }
    out[0] = acc[5];
    out[1] = acc[6];
}

// 2-limb x 2-limb -> 4-limb
__host__ __device__ inline void glv_mul128(const u64 a[2], const u64 b[2], u64 out[4]) {
    unsigned __int128 t0 = (unsigned __int128)a[0] * b[0];
    unsigned __int128 t1 = (unsigned __int128)a[0] * b[1];
    unsigned __int128 t2 = (unsigned __int128)a[1] * b[0];
    unsigned __int128 t3 = (unsigned __int128)a[1] * b[1];
    out[0] = (u64)t0;
    unsigned __int128 m = (t0 >> 64) + (u64)t1 + (u64)t2;
    out[1] = (u64)m;
    unsigned __int128 h = (m >> 64) + (t1 >> 64) + (t2 >> 64) + (u64)t3;
    out[2] = (u64)h;
    out[3] = (u64)(h >> 64) + (u64)(t3 >> 64);
}

__host__ __device__ inline int glv_cmp4(const u64 a[4], const u64 b[4]) {
    for (int i = 3; i >= 0; --i) {
        if (a[i] != b[i]) return a[i] > b[i] ? 1 : -1;
    }
    return 0;
}
__host__ __device__ inline void glv_sub4(const u64 a[4], const u64 b[4], u64 out[4]) {
    unsigned __int128 borrow = 0;
    for (int i = 0; i < 4; ++i) {
        unsigned __int128 t = (unsigned __int128)a[i] - b[i] - (u64)borrow;
        out[i] = (u64)t;
        borrow = (t >> 64) ? 1 : 0;
    }
}
__host__ __device__ inline void glv_add4(const u64 a[4], const u64 b[4], u64 out[4]) {
    unsigned __int128 carry = 0;
    for (int i = 0; i < 4; ++i) {
        unsigned __int128 t = (unsigned __int128)a[i] + b[i] + (u64)carry;
        out[i] = (u64)t;
        carry = t >> 64;
    }
}

// canonical 4-limb scalars -> per scalar 4 u64: k1 lo, k1 hi|sign<<63,
// k2 lo, k2 hi|sign<<63 (magnitudes < 2^127)
__global__ __launch_bounds__(256) void k_glv_decompose(const u64* canon,
                                                       uint32_t count, u64* out) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= count) return;
    const u64 G1C[4] = GLV_G1;
    const u64 G2C[4] = GLV_G2;
    const u64 A1[2] = GLV_A1;
    const u64 B1[2] = GLV_B1;  // |b1|
    const u64 A2[2] = GLV_A2;
    const u64 B2[2] = GLV_B2;
    u64 k[4] = {canon[4 * i], canon[4 * i + 1], canon[4 * i + 2], canon[4 * i + 3]};
    u64 c1[2], c2[2];
    glv_mul_shift(G1C, k, c1);
    glv_mul_shift(G2C, k, c2);
    // k1 = k - c1*a1 - c2*a2  (signed)
    u64 s1[4], s2[4], S[4];
    glv_mul128(c1, A1, s1);
    glv_mul128(c2, A2, s2);
    glv_add4(s1, s2, S);
    u64 k1[4];
    u64 sign1;
    if (glv_cmp4(k, S) >= 0) {
        glv_sub4(k, S, k1);
        sign1 = 0;
    } else {
        glv_sub4(S, k, k1);
        sign1 = 1;
    }
    // k2 = c1*|b1| - c2*b2  (signed)
    u64 t1[4], t2[4], k2[4];
    u64 sign2;
    glv_mul128(c1, B1, t1);
    glv_mul128(c2, B2, t2);
    if (glv_cmp4(t1, t2) >= 0) {
        glv_sub4(t1, t2, k2);
        sign2 = 0;
    } else {
        glv_sub4(t2, t1, k2);
        sign2 = 1;
    }
    out[4 * i] = k1[0];
    out[4 * i + 1] = k1[1] | (sign1 << 63);
    out[4 * i + 2] = k2[0];
    out[4 * i + 3] = k2[1] | (sign2 << 63);
}

// interleave bases with their endomorphism images: out[2i] = P_i,
// out[2i+1] = phi(P_i) = (beta*x, y) — adjacency keeps the bucket-phase
// gathers cache-local (a split phi array costs ~15% on the 2^20 MSM)
__global__ __launch_bounds__(256) void k_bases_interleave(const G1Aff* in, G1Aff* out,
                                                          uint32_t count) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= count) return;
    const u64 BETA[4] = GLV_BETA_MONT;
    Fq beta;
    beta.l[0] = BETA[0];
    beta.l[1] = BETA[1];
    beta.l[2] = BETA[2];
    beta.l[3] = BETA[3];
    G1Aff p = in[i];
    out[2 * i] = p;
    p.x = p.x.mul(beta);
    out[2 * i + 1] = p;
}

// GLV digit decomposition: each scalar contributes TWO 127-bit halves whose
// digits share the (poly, window) bucket space; half 2 gathers phi(bases)
// via bit 30 of the val record.
__global__ __launch_bounds__(256) void k_msm_digits_glv(const u64* glv, uint32_t n,
                                                        uint32_t c, uint32_t W,
                                                        uint32_t B, uint32_t* keys,
                                                        uint32_t* vals) {
    uint32_t idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n * B) return;
    uint32_t b = idx / n, i = idx % n;
    uint32_t half = 1u << (c - 1);
    uint64_t cmask = (1ull << c) - 1;
    for (uint32_t h = 0; h < 2; ++h) {
        u64 s0 = glv[4 * idx + 2 * h];
        u64 s1raw = glv[4 * idx + 2 * h + 1];
        uint32_t hsign = (uint32_t)(s1raw >> 63);
        u64 s1 = s1raw & ((1ull << 63) - 1);
        uint32_t carry = 0;
        for (uint32_t w = 0; w < W; ++w) {
            uint32_t bit0 = w * c;
            u64 raw;
            if (bit0 < 64) {
                raw = s0 >> bit0;
                if (bit0 + c > 64) raw |= s1 << (64 - bit0);
            } else {
                raw = s1 >> (bit0 - 64);
            }
            raw = (raw & cmask) + carry;
            uint32_t mag, sign;
            if (raw >= half) {
                if (raw > half) {
                    mag = (uint32_t)((1ull << c) - raw);
                    sign = 1;
                    carry = 1;
                } else {
                    mag = half;
                    sign = 0;
                    carry = 0;
                }
            } else {
                mag = (uint32_t)raw;
                sign = 0;
                carry = 0;
            }
            uint64_t o = ((uint64_t)(b * W + w)) * (2ull * n) + (uint64_t)h * n + i;
            keys[o] = mag == 0 ? msm_sentinel(B * W) : (((b * W + w) << 16) | mag);
            vals[o] = ((sign ^ hsign) << 31) | (2 * i + h);
        }
    }
}

// ---- 3a. segment-head flags (for stream compaction) ----
__global__ void k_msm_head_flags(const uint32_t* keys, uint32_t total, uint32_t sentinel,
                                 uint8_t* flags) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= total) return;
    uint32_t key = keys[t];
    flags[t] = (key != sentinel && (t == 0 || keys[t - 1] != key)) ? 1 : 0;
}

// Segment walks are capped at MSM_MAX_SEG entries: skewed digit
// distributions (the top window of <2^253 scalars, duplicate scalars) create
// segments thousands of entries long, and a single lane walking one is the
// whole kernel's critical path.  Long segments are split into sub-segments
// reduced in parallel, then merged per segment.
constexpr uint32_t MSM_MAX_SEG = 128;

// choose the sub-segment cap so the reduce kernel has >= ~64K lanes in
// flight (256 CUs want >> 256 workgroups); smaller caps cost extra merge
// work, so keep within [8, MSM_MAX_SEG]
__host__ inline uint32_t msm_seg_cap(uint64_t total_entries) {
    uint64_t cap = total_entries / (1u << 16);
    if (cap < 8) cap = 8;  // scratch sizes subs as total/8 — keep min 8
    if (cap > MSM_MAX_SEG) cap = MSM_MAX_SEG;
    return (uint32_t)cap;
}

// ---- 3b. segment lengths + sub-segment counts ----
// heads are in increasing order (rocprim::select is stable); seg i spans
// [heads[i], heads[i+1] or first sentinel/total).
__global__ void k_msm_seg_lengths(const uint32_t* keys, const uint32_t* heads,
                                  const uint32_t* head_count, uint32_t total,
                                  uint32_t* lens, uint32_t* nsub, uint32_t seg_cap) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    uint32_t hc = *head_count;
    if (t >= hc) return;
    uint32_t start = heads[t];
    uint32_t end = (t + 1 < hc) ? heads[t + 1] : total;
    // sentinel tail: the zero-digit entries sort after every real key, and
    // the last real segment's end must exclude them
    if (t + 1 == hc) {
        uint32_t key = keys[start];
        uint32_t e = start;
        while (e < total && keys[e] == key) ++e;
        end = e;
    }
    uint32_t len = end - start;
    lens[t] = len;
    nsub[t] = (len + seg_cap - 1) / seg_cap;
}

// ---- 3c. emit sub-segment records (after exclusive scan of nsub) ----
// subs: per sub-segment (start, len, head index)
__global__ void k_msm_make_subs(const uint32_t* heads, const uint32_t* lens,
                                const uint32_t* sub_off, const uint32_t* head_count,
                                uint32_t* sub_start, uint32_t* sub_len,
                                uint32_t seg_cap) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= *head_count) return;
    uint32_t start = heads[t], len = lens[t], off = sub_off[t];
    uint32_t k = 0;
    while (len > 0) {
        uint32_t l = len < seg_cap ? len : seg_cap;
        sub_start[off + k] = start;
        sub_len[off + k] = l;
        start += l;
        len -= l;
        ++k;
    }
}

// ---- 3d. sub-segment reduction: one thread per sub-segment (<= MAX_SEG) ----
__global__ __launch_bounds__(256) void k_msm_bucket_reduce(
    const uint32_t* vals, const uint32_t* sub_start, const uint32_t* sub_len,
    const uint32_t* sub_order /* sub ids sorted by length desc */,
    uint32_t sub_count, const G1Aff* bases, G1Jac* partials2) {
    uint32_t tt = blockIdx.x * blockDim.x + threadIdx.x;
    if (tt >= sub_count) return;
    uint32_t t = sub_order[tt];
    uint32_t start = sub_start[t], len = sub_len[t];
    // NOTE on ILP: a two-accumulator interleave was tried and reverted —
    // the v_mad_u64_u32 hazard s_nops it targets come from VCC carry
    // serialization, which BOTH chains share, so it only added an extra EC
    // add per sub-segment; the hazard slots are instead covered by running
    // 16 concurrent proof streams.
    G1Jac acc = G1Jac::identity();
    for (uint32_t j = start; j < start + len; ++j) {
        uint32_t v = vals[j];
        G1Aff p = bases[v & 0x7FFFFFFFu];
        acc = acc.madd(p, (v >> 31) != 0);
    }
    partials2[t] = acc;
}

// ---- 3e. merge sub-partials per segment -> bucket ----
// buckets: W * 2^(c-1) Jacobian points (zero-init = identity for untouched).
__global__ __launch_bounds__(256) void k_msm_seg_merge(
    const uint32_t* keys, const uint32_t* heads, const uint32_t* sub_off,
    const uint32_t* nsub, const uint32_t* head_order /* by nsub desc */,
    uint32_t head_count, const G1Jac* partials2,
    G1Jac* buckets, uint32_t c) {
    uint32_t tt = blockIdx.x * blockDim.x + threadIdx.x;
    if (tt >= head_count) return;
    uint32_t t = head_order[tt];
    uint32_t off = sub_off[t], ns = nsub[t];
    G1Jac acc = partials2[off];
    for (uint32_t k = 1; k < ns; ++k) acc = acc.add(partials2[off + k]);
    uint32_t key = keys[heads[t]];
    uint32_t w = key >> 16;
    uint32_t mag = key & 0xFFFFu;  // 1 .. 2^(c-1)
    buckets[(uint64_t)w * (1u << (c - 1)) + (mag - 1)] = acc;
}

// ---- 4. per-window chunked suffix sums ----
// grid: W * (2^(c-1) / MSM_CHUNK) threads total; partials: per thread
// (T = plain sum, S = locally-weighted sum) -> 2 Jacobians.
__global__ __launch_bounds__(256) void k_msm_window_chunks(const G1Jac* buckets, uint32_t c, uint32_t W,
                                    uint32_t chunk_sz,
                                    G1Jac* partials /* 2 per thread: T, S */) {
    uint32_t nb = 1u << (c - 1);
    uint32_t chunks_per_w = nb / chunk_sz;
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= W * chunks_per_w) return;
    uint32_t w = t / chunks_per_w;
    uint32_t chunk = t % chunks_per_w;
    const G1Jac* b = buckets + (uint64_t)w * nb + (uint64_t)chunk * chunk_sz;
    // local digits are base + j, j = 1..chunk_sz, base = chunk*chunk_sz
    // suffix running sums over j descending: run = sum_{k>=j} S_k ; S += run
    G1Jac run = G1Jac::identity(), S = G1Jac::identity();
    for (int j = (int)chunk_sz - 1; j >= 0; --j) {
        run = run.add(b[j]);
        S = S.add(run);
    }
    partials[2 * t] = run;  // T = plain sum of chunk buckets
    partials[2 * t + 1] = S;  // sum_j (j_local) * bucket, j_local = 1..chunk_sz
}

// ---- 4b. fused window fold (small windows): one block per key group,
// one lane per chunk, no per-chunk base weighting.  Uses the identity
//   sum_j (j+1) B_j = sum_l S_l + chunk_sz * sum_{l>=1} SufT_l
// with SufT_l = sum_{k>=l} T_k computed by an LDS suffix scan; replaces
// k_msm_window_chunks + k_msm_window_combine when chunks_per_window <= 256
// (~1.8x less EC work and one kernel + HBM round trip fewer).
__global__ __launch_bounds__(256) void k_msm_window_fold(const G1Jac* buckets,
                                                         uint32_t c,
                                                         uint32_t chunk_sz,
                                                         G1Jac* window_out) {
    __shared__ G1Jac sA[256], sB[256];
    uint32_t g = blockIdx.x;
    uint32_t nb = 1u << (c - 1);
    uint32_t L = nb / chunk_sz;  // == blockDim.x
    uint32_t l = threadIdx.x;
    const G1Jac* b = buckets + (uint64_t)g * nb + (uint64_t)l * chunk_sz;
    G1Jac T = G1Jac::identity(), S = G1Jac::identity();
    for (int j = (int)chunk_sz - 1; j >= 0; --j) {
        T = T.add(b[j]);
        S = S.add(T);
    }
    // suffix scan of T (ping-pong)
    G1Jac* cur = sA;
    G1Jac* nxt = sB;
    cur[l] = T;
    __syncthreads();
    for (uint32_t st = 1; st < L; st <<= 1) {
        G1Jac v = cur[l];
        if (l + st < L) v = v.add(cur[l + st]);
        nxt[l] = v;
        __syncthreads();
        G1Jac* t2 = cur;
        cur = nxt;
        nxt = t2;
    }
    // pair tree: U = sum S_l (in nxt), V = sum_{l>=1} SufT_l (in cur)
    G1Jac vl = (l >= 1) ? cur[l] : G1Jac::identity();
    __syncthreads();
    cur[l] = vl;
    nxt[l] = S;
    __syncthreads();
    for (uint32_t st = L / 2; st > 0; st >>= 1) {
        if (l < st) {
            nxt[l] = nxt[l].add(nxt[l + st]);
            cur[l] = cur[l].add(cur[l + st]);
        }
        __syncthreads();
    }
    if (l == 0) {
        G1Jac V = cur[0];
        for (uint32_t m = chunk_sz; m > 1; m >>= 1) V = V.dbl();
        window_out[g] = nxt[0].add(V);
    }
}

// ---- 5a. window combine: grid = W * MSM_SUBB blocks; block (w, sb) folds a
// slice of window w's chunk partials (contribution = S + base*T with
// base = chunk*CHUNK), LDS tree reduce -> window_partials[w*SUBB + sb].
constexpr uint32_t MSM_SUBB = 16;

__global__ __launch_bounds__(64) void k_msm_window_combine(const G1Jac* partials,
                                                           uint32_t c,
                                                           uint32_t chunk_sz,
                                                           uint32_t subb,
                                                           G1Jac* window_partials) {
    __shared__ G1Jac red[64];
    uint32_t w = blockIdx.x / subb;
    uint32_t sb = blockIdx.x % subb;
    uint32_t nb = 1u << (c - 1);
    uint32_t chunks_per_w = nb / chunk_sz;
    uint32_t per_sb = (chunks_per_w + subb - 1) / subb;
    uint32_t lo = sb * per_sb;
    uint32_t hi = lo + per_sb < chunks_per_w ? lo + per_sb : chunks_per_w;
    G1Jac acc = G1Jac::identity();
    for (uint32_t chunk = lo + threadIdx.x; chunk < hi; chunk += blockDim.x) {
        uint32_t t = w * chunks_per_w + chunk;
        G1Jac T = partials[2 * t];
        G1Jac S = partials[2 * t + 1];
        uint32_t base = chunk * chunk_sz;
        G1Jac bT = G1Jac::identity();
        G1Jac addend = T;
        while (base) {
            if (base & 1) bT = bT.add(addend);
            addend = addend.dbl();
            base >>= 1;
        }
        acc = acc.add(S).add(bT);
    }
    red[threadIdx.x] = acc;
    __syncthreads();
    for (uint32_t stride = blockDim.x / 2; stride > 0; stride >>= 1) {
        if (threadIdx.x < stride)
            red[threadIdx.x] = red[threadIdx.x].add(red[threadIdx.x + stride]);
        __syncthreads();
    }
    if (threadIdx.x == 0) window_partials[w * subb + sb] = red[0];
}

// ---- 5b. scan-based combine (large windows): block (w, sb) folds P =
// chunks_per_window/subb chunk partials with NO per-chunk scalar weighting.
// Decompose the weighted sum over the block's chunk range [lo, lo+P):
//   sum_c (S_c + cz*c*T_c) = sum S + cz*( lo*At + pl*sum_l l*A_l
//                                         + sum_l (B_l - A_l) )
// where lane l serially folds pl = P/64 chunks into A_l = sum T,
// B_l = sum (j+1)*T (suffix-running trick), At = sum_l A_l, and
// sum_l l*A_l comes from one LDS suffix scan.  The only scalar weights
// left are the power-of-two lo and cz (a few doublings once per block) —
// replaces the per-chunk binary-doubling chains of k_msm_window_combine,
// which rocprof showed at 1.74 ms of the 6.2 ms 2^20 MSM.
__global__ __launch_bounds__(64) void k_msm_window_combine2(
    const G1Jac* partials, uint32_t c, uint32_t chunk_sz, uint32_t subb,
    G1Jac* window_partials) {
    __shared__ G1Jac sA[64], sB[64], sS[64];
    uint32_t w = blockIdx.x / subb;
    uint32_t sb = blockIdx.x % subb;
    uint32_t nb = 1u << (c - 1);
    uint32_t chunks_per_w = nb / chunk_sz;
    uint32_t P = chunks_per_w / subb;  // caller guarantees P >= 64
    uint32_t pl = P / 64;
    uint32_t lo = sb * P;
    uint32_t l = threadIdx.x;
    const G1Jac* base = partials + 2 * ((uint64_t)w * chunks_per_w + lo + l * pl);
    G1Jac A = G1Jac::identity(), B = G1Jac::identity(), S = G1Jac::identity();
    for (int j = (int)pl - 1; j >= 0; --j) {
        A = A.add(base[2 * j]);        // run += T_j
        B = B.add(A);                  // B = sum (j+1)*T_j
        S = S.add(base[2 * j + 1]);
    }
    // LDS suffix scan of A -> SufA_l = sum_{m>=l} A_m (ping-pong not needed:
    // Hillis-Steele with double buffer via barrier pairs)
    sA[l] = A;
    sB[l] = B;
    sS[l] = S;
    __syncthreads();
    for (uint32_t st = 1; st < 64; st <<= 1) {
        G1Jac v = sA[l];
        if (l + st < 64) v = v.add(sA[l + st]);
        __syncthreads();
        sA[l] = v;
        __syncthreads();
    }
    // At = SufA_0 (visible to all lanes); parallel trees for
    // lw = sum_{l>=1} SufA_l, Bt = sum B_l, St = sum S_l
    G1Jac At = sA[0];
    G1Jac tail = (l >= 1) ? sA[l] : G1Jac::identity();
    __syncthreads();
    sA[l] = tail;
    __syncthreads();
    for (uint32_t st = 32; st > 0; st >>= 1) {
        if (l < st) {
            sA[l] = sA[l].add(sA[l + st]);
            sB[l] = sB[l].add(sB[l + st]);
            sS[l] = sS[l].add(sS[l + st]);
        }
        __syncthreads();
    }
    if (l == 0) {
        G1Jac lw = sA[0], Bt = sB[0], St = sS[0];
        // weighted = lo*At + pl*lw + (Bt - At); lo has <=2 set bits
        G1Jac loAt = G1Jac::identity();
        {
            G1Jac addend = At;
            for (uint32_t rem = lo; rem; rem >>= 1) {
                if (rem & 1) loAt = loAt.add(addend);
                addend = addend.dbl();
            }
        }
        G1Jac plw = lw;
        for (uint32_t m = pl; m > 1; m >>= 1) plw = plw.dbl();
        G1Jac nAt = At;
        nAt.Y = nAt.Y.neg();  // Bt - At
        G1Jac weighted = loAt.add(plw).add(Bt).add(nAt);
        for (uint32_t m = chunk_sz; m > 1; m >>= 1) weighted = weighted.dbl();
        window_partials[blockIdx.x] = St.add(weighted);
    }
}

// (The final fold across windows — W*SUBB <= 512 Jacobians, ~1.5 KB — is
// done on the HOST: a single-lane dependent EC chain runs ~50x slower on a
// GPU SIMT lane than on a host core, and the data is tiny.)

// ---- helpers ----
__global__ void k_fr_to_canonical(const Fr* in, uint64_t* out, uint32_t n) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    in[i].to_canonical(out + 4 * i);
}

}  // namespace rng
