// PRODUCT PATH — MI355X-native Pippenger MSM over BN254 G1.
//
// Replaces: arkworks ark-ec VariableBaseMSM as consumed through the
// reference's KZG commitments (BASELINE config #2).
//
// Structure (sort-based, designed for gfx950):
//  1. k_msm_digits: signed windowed digit decomposition (c bits, digits in
//     [-2^(c-1), 2^(c-1)]), one (key = window<<16 | magnitude, val =
//     sign<<31 | point index) pair per nonzero digit; zero digits get a
//     sentinel key that sorts last.
//  2. rocPRIM device radix sort on the 21-bit keys (groups all points of a
//     bucket together — no atomics or EC critical sections anywhere).
//  3. k_msm_bucket_reduce: one thread per segment head walks its bucket's
//     points with Jacobian mixed-adds (bases gathered from HBM).
//  4. k_msm_window_chunks: per (window, chunk of 2^(c-1)/CHUNK buckets):
//     suffix running sums -> (sum, weighted-sum) partials.
//  5. k_msm_final: combine chunk partials, fold windows (Horner with c
//     doublings), single workgroup; result Jacobian to host.
#include <hip/hip_runtime.h>
#include <rocprim/device/device_radix_sort.hpp>
#include <rocprim/device/device_select.hpp>
#include <rocprim/device/device_scan.hpp>
#include <rocprim/iterator/counting_iterator.hpp>
#include "gpu_curve.hpp"

namespace rng {

constexpr uint32_t MSM_SENTINEL = 0x3FFFFFFu;  // > any (group<<16|mag); 26 bits
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
        keys[o] = mag == 0 ? MSM_SENTINEL : (((b * W + w) << 16) | mag);
        vals[o] = (sign << 31) | i;
    }
    // carry out of the top window must be zero for scalars < 2^(W*c-1)
}

// ---- 3a. segment-head flags (for stream compaction) ----
__global__ void k_msm_head_flags(const uint32_t* keys, uint32_t total, uint8_t* flags) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= total) return;
    uint32_t key = keys[t];
    flags[t] = (key != MSM_SENTINEL && (t == 0 || keys[t - 1] != key)) ? 1 : 0;
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
    if (cap < 8) cap = 8;
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
    const uint32_t* nsub, uint32_t head_count, const G1Jac* partials2,
    G1Jac* buckets, uint32_t c) {
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= head_count) return;
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
