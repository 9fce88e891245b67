// PRODUCT PATH — MI355X-native radix-2 NTT over BN254 Fr.
//
// Replaces: arkworks ark-poly Radix2EvaluationDomain fft/ifft as consumed by
// the reference prover (BASELINE config #3).  Natural order
// in and out, matching the convention the CPU oracle pins (oracle/ntt.hpp).
//
// Structure (designed for gfx950, not a port):
//  - n <= 4096:  one workgroup per transform, whole transform staged in LDS
//    (4096 * 32 B = 128 KiB), DIT butterflies after a bit-reversed load.
//  - n >= 8192:  four-step N = N1*N2 decomposition, two passes:
//      K1: column pairs (stride N1) -> length-N2 DFT in LDS -> outer twiddle
//          w_N^(j1*k2) from two product tables (TA/TB, L2-resident) -> store.
//      K2: row pairs (contiguous)   -> length-N1 DFT in LDS -> strided store
//          to k2 + N2*k1 (natural order out).
//    Column/row PAIRS make every global access a 64-byte aligned segment
//    (2 adjacent Fr of 32 B), the HBM sector size — full-bandwidth pattern.
//  - LDS element indices are XOR-swizzled so power-of-two-strided butterfly
//    accesses spread across the 64 LDS banks.
// All twiddles are precomputed per (n, direction) into HBM once per plan.
#include <hip/hip_runtime.h>
#include "gpu_field.hpp"

namespace rng {

// XOR swizzle on the low 3 bits with a fold of higher bits: keeps
// power-of-two-strided element accesses off a single bank group.
__device__ __forceinline__ uint32_t lds_slot(uint32_t i) {
    return i ^ (((i >> 3) ^ (i >> 6) ^ (i >> 9) ^ (i >> 12)) & 7u);
}

__device__ __forceinline__ uint32_t brev_n(uint32_t x, uint32_t logn) {
    return __brev(x) >> (32 - logn);
}

// ---- twiddle table generation ----
// table[i] = base^(i*step) for i < count (Montgomery in/out).
__global__ void k_pow_table(Fr* out, Fr base, uint64_t step, uint64_t count) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= count) return;
    uint64_t e = i * step;
    out[i] = base.pow_u64(e);
}

// ---- single-workgroup NTT (n <= 4096), one transform per block ----
__global__ __launch_bounds__(512) void k_ntt_small(Fr* data, const Fr* wst,
                                                   uint32_t n, uint32_t logn,
                                                   Fr scale, int do_scale) {
    extern __shared__ Fr lds[];
    Fr* base = data + (uint64_t)blockIdx.x * n;
    for (uint32_t q = threadIdx.x; q < n; q += blockDim.x)
        lds[lds_slot(brev_n(q, logn))] = base[q];
    __syncthreads();
    for (uint32_t s = 1; s <= logn; ++s) {
        uint32_t half = 1u << (s - 1);
        for (uint32_t q = threadIdx.x; q < n / 2; q += blockDim.x) {
            uint32_t blk = q >> (s - 1);
            uint32_t k = q & (half - 1);
            uint32_t i0 = (blk << s) + k;
            uint32_t i1 = i0 + half;
            Fr tw = wst[k << (logn - s)];
            Fr u = lds[lds_slot(i0)];
            Fr v = lds[lds_slot(i1)].mul(tw);
            lds[lds_slot(i0)] = u.add(v);
            lds[lds_slot(i1)] = u.sub(v);
        }
        __syncthreads();
    }
    for (uint32_t q = threadIdx.x; q < n; q += blockDim.x) {
        Fr v = lds[lds_slot(q)];
        if (do_scale) v = v.mul(scale);
        base[q] = v;
    }
}

// ---- pass 1: column DFT of length N2 (stride N1) + outer twiddle ----
// grid.x = N1/PW * batch; data viewed as [N2 rows][N1 cols].  PW = columns
// per block: 2 makes every global access a 64-byte segment (2 adjacent Fr),
// 1 halves the LDS footprint.  At 2^22 the pair form needs 128 KiB LDS =
// ONE block per CU (2 waves/SIMD) and the dependent butterfly muls run
// latency-exposed; singles fit 2 blocks/CU and trade half-sector global
// access (~0.05 ms of traffic) for doubled occupancy.
template <int PW>
__global__ __launch_bounds__(512) void k_ntt_col(Fr* data, const Fr* wst2,
                                                 const Fr* ta, const Fr* tb,
                                                 uint32_t N1, uint32_t N2,
                                                 uint32_t logN2, uint32_t split_log) {
    extern __shared__ Fr lds[];
    uint32_t wg = blockIdx.x % (N1 / PW);
    Fr* base = data + (uint64_t)(blockIdx.x / (N1 / PW)) * N1 * N2;
    uint32_t split_mask = (1u << split_log) - 1;

    for (uint32_t q = threadIdx.x; q < PW * N2; q += blockDim.x) {
        uint32_t m = q / PW, c = q % PW;
        Fr v = base[(PW * wg + c) + (uint64_t)N1 * m];
        lds[lds_slot(c * N2 + brev_n(m, logN2))] = v;
    }
    __syncthreads();
    for (uint32_t s = 1; s <= logN2; ++s) {
        uint32_t half = 1u << (s - 1);
        for (uint32_t q = threadIdx.x; q < PW * (N2 / 2); q += blockDim.x) {
            uint32_t c = q / (N2 / 2);
            uint32_t bf = q & (N2 / 2 - 1);
            uint32_t blk = bf >> (s - 1);
            uint32_t k = bf & (half - 1);
            uint32_t i0 = c * N2 + (blk << s) + k;
            uint32_t i1 = i0 + half;
            Fr tw = wst2[k << (logN2 - s)];
            Fr u = lds[lds_slot(i0)];
            Fr v = lds[lds_slot(i1)].mul(tw);
            lds[lds_slot(i0)] = u.add(v);
            lds[lds_slot(i1)] = u.sub(v);
        }
        __syncthreads();
    }
    for (uint32_t q = threadIdx.x; q < PW * N2; q += blockDim.x) {
        uint32_t k2 = q / PW, c = q % PW;
        uint32_t j1 = PW * wg + c;
        Fr tw = ta[(uint64_t)j1 * (k2 >> split_log)].mul(tb[(uint64_t)j1 * (k2 & split_mask)]);
        base[j1 + (uint64_t)N1 * k2] = lds[lds_slot(c * N2 + k2)].mul(tw);
    }
}

// ---- pass 2: row DFT of length N1 (contiguous) + strided store ----
// grid.x = N2/PW * batch.
template <int PW>
__global__ __launch_bounds__(512) void k_ntt_row(const Fr* in, Fr* out, const Fr* wst1,
                                                 uint32_t N1, uint32_t N2,
                                                 uint32_t logN1, Fr scale, int do_scale) {
    extern __shared__ Fr lds[];
    uint32_t wg = blockIdx.x % (N2 / PW);
    uint64_t boff = (uint64_t)(blockIdx.x / (N2 / PW)) * N1 * N2;
    const Fr* ibase = in + boff;
    Fr* obase = out + boff;

    for (uint32_t q = threadIdx.x; q < PW * N1; q += blockDim.x) {
        uint32_t c = q >> logN1;
        uint32_t j1 = q & (N1 - 1);
        Fr v = ibase[(uint64_t)(PW * wg) * N1 + q];
        lds[lds_slot(c * N1 + brev_n(j1, logN1))] = v;
    }
    __syncthreads();
    for (uint32_t s = 1; s <= logN1; ++s) {
        uint32_t half = 1u << (s - 1);
        for (uint32_t q = threadIdx.x; q < PW * (N1 / 2); q += blockDim.x) {
            uint32_t c = q / (N1 / 2);
            uint32_t bf = q & (N1 / 2 - 1);
            uint32_t blk = bf >> (s - 1);
            uint32_t k = bf & (half - 1);
            uint32_t i0 = c * N1 + (blk << s) + k;
            uint32_t i1 = i0 + half;
            Fr tw = wst1[k << (logN1 - s)];
            Fr u = lds[lds_slot(i0)];
            Fr v = lds[lds_slot(i1)].mul(tw);
            lds[lds_slot(i0)] = u.add(v);
            lds[lds_slot(i1)] = u.sub(v);
        }
        __syncthreads();
    }
    // element [c][k1] -> global position k2 + N2*k1, k2 = PW*wg + c
    for (uint32_t q = threadIdx.x; q < PW * N1; q += blockDim.x) {
        uint32_t k1 = q / PW, c = q % PW;
        Fr v = lds[lds_slot(c * N1 + k1)];
        if (do_scale) v = v.mul(scale);
        obase[(PW * wg + c) + (uint64_t)N2 * k1] = v;
    }
}

}  // namespace rng
