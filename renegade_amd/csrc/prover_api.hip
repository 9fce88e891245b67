// PRODUCT PATH — C-ABI implementation (include/rng_prover.h).
//
// Host orchestration for the MI355X prover backend: SRS load (ptau
// semantics of crates/circuits/circuit-types/src/primitives/srs.rs:63-214),
// NTT plans, MSM pipeline.  Single translation unit: includes the kernel
// files directly.
#include <cstdio>
#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <vector>

#include "../../include/rng_prover.h"
#include "ntt_kernels.hip"
#include "msm_kernels.hip"
#include "plonk_circuit.hpp"
#include "test_circuits.hpp"

namespace rng {

#define HIP_CHECK(x)                                                   \
    do {                                                               \
        hipError_t err__ = (x);                                        \
        if (err__ != hipSuccess) {                                     \
            fprintf(stderr, "rng_prover: HIP error %s at %s:%d\n",     \
                    hipGetErrorString(err__), __FILE__, __LINE__);     \
            return RNG_ERR_HIP;                                        \
        }                                                              \
    } while (0)

static bool gpu_ok() {
    static int cached = -1;
    if (cached < 0) {
        int n = 0;
        cached = (hipGetDeviceCount(&n) == hipSuccess && n > 0) ? 1 : 0;
    }
    return cached == 1;
}

static thread_local double tls_msm_times[5] = {0, 0, 0, 0, 0};
static thread_local double tls_ntt_times[2] = {0, 0};

struct EvtTimer {
    hipEvent_t ev[8];
    int used = 0;
    EvtTimer() {
        for (auto& e : ev) hipEventCreate(&e);
    }
    ~EvtTimer() {
        for (auto& e : ev) hipEventDestroy(e);
    }
    void mark(hipStream_t s) {
        if (used < 8) hipEventRecord(ev[used++], s);
    }
    void collect(double* out, int n) {
        hipEventSynchronize(ev[used - 1]);
        for (int i = 0; i < n && i + 1 < used; ++i) {
            float ms = 0;
            hipEventElapsedTime(&ms, ev[i], ev[i + 1]);
            out[i] = ms;
        }
    }
};

// ---------------- NTT plans ----------------

struct NttPlan {
    uint32_t n, logn, N1, N2, logN1, logN2, split_log;
    // device tables (fwd, inv): stage tables for len N1 and N2, outer TA/TB
    Fr *wst1_f = nullptr, *wst2_f = nullptr, *ta_f = nullptr, *tb_f = nullptr;
    Fr *wst1_i = nullptr, *wst2_i = nullptr, *ta_i = nullptr, *tb_i = nullptr;
    Fr ninv;  // Montgomery form
    Fr* scratch = nullptr;  // n elements, for the pass-2 ping-pong (resized w/ batch)
    uint64_t scratch_elems = 0;
};

// host-side helpers over the shared field type (host-compiled path of Fp4)
static Fr h_fr_root_of_unity(uint32_t n) {
    static const u64 root_c[4] = FR_TWO_ADIC_ROOT;
    Fr root = Fr::from_canonical(root_c);
    uint32_t logn = 0;
    while ((1u << logn) < n) logn++;
    for (uint32_t i = logn; i < FR_TWO_ADICITY; ++i) root = root.sqr();
    return root;
}

struct RngCtxImpl {
    // SRS
    std::vector<uint64_t> srs_g1_host;  // (max_degree+1) * 8 u64 packed affine
    uint64_t srs_count = 0;
    void* srs_dev = nullptr;
    uint64_t h_g2[16];       // h: x.c0,x.c1,y.c0,y.c1 Montgomery
    uint64_t beta_h_g2[16];
    std::map<uint32_t, std::unique_ptr<NttPlan>> plans;
    std::mutex mu;

    ~RngCtxImpl() {
        for (auto& kv : plans) {
            NttPlan* p = kv.second.get();
            for (Fr* b : {p->wst1_f, p->wst2_f, p->ta_f, p->tb_f, p->wst1_i,
                          p->wst2_i, p->ta_i, p->tb_i, p->scratch})
                if (b) hipFree(b);
        }
        if (srs_dev) hipFree(srs_dev);
    }
};

static int build_pow_table(Fr** out, const Fr& base, uint64_t step, uint64_t count) {
    HIP_CHECK(hipMalloc(out, count * sizeof(Fr)));
    uint32_t blocks = (uint32_t)((count + 255) / 256);
    hipLaunchKernelGGL(k_pow_table, dim3(blocks), dim3(256), 0, 0, *out, base, step, count);
    HIP_CHECK(hipGetLastError());
    return RNG_OK;
}

static NttPlan* get_plan(RngCtxImpl* ctx, uint32_t n, uint64_t batch) {
    std::lock_guard<std::mutex> lk(ctx->mu);
    auto it = ctx->plans.find(n);
    NttPlan* p;
    if (it != ctx->plans.end()) {
        p = it->second.get();
    } else {
        auto np = std::make_unique<NttPlan>();
        p = np.get();
        p->n = n;
        p->logn = 0;
        while ((1u << p->logn) < n) p->logn++;
        p->logN2 = p->logn / 2;
        p->logN1 = p->logn - p->logN2;
        p->N1 = 1u << p->logN1;
        p->N2 = 1u << p->logN2;
        p->split_log = p->logN2 < 6 ? p->logN2 : 6;
        Fr w = h_fr_root_of_unity(n);
        Fr wi = w.inverse();
        Fr wN1_f = w.pow_u64(p->N2), wN2_f = w.pow_u64(p->N1);
        Fr wN1_i = wi.pow_u64(p->N2), wN2_i = wi.pow_u64(p->N1);
        uint64_t split = 1ull << p->split_log;
        if (n <= 4096) {
            // single-WG path needs only the full-length stage table (in wst1)
            if (build_pow_table(&p->wst1_f, w, 1, n / 2) != RNG_OK) return nullptr;
            if (build_pow_table(&p->wst1_i, wi, 1, n / 2) != RNG_OK) return nullptr;
        } else {
            if (build_pow_table(&p->wst1_f, wN1_f, 1, p->N1 / 2) != RNG_OK) return nullptr;
            if (build_pow_table(&p->wst2_f, wN2_f, 1, p->N2 / 2) != RNG_OK) return nullptr;
            if (build_pow_table(&p->ta_f, w, split, (uint64_t)p->N1 * (p->N2 >> p->split_log)) != RNG_OK) return nullptr;
            if (build_pow_table(&p->tb_f, w, 1, (uint64_t)p->N1 * split) != RNG_OK) return nullptr;
            if (build_pow_table(&p->wst1_i, wN1_i, 1, p->N1 / 2) != RNG_OK) return nullptr;
            if (build_pow_table(&p->wst2_i, wN2_i, 1, p->N2 / 2) != RNG_OK) return nullptr;
            if (build_pow_table(&p->ta_i, wi, split, (uint64_t)p->N1 * (p->N2 >> p->split_log)) != RNG_OK) return nullptr;
            if (build_pow_table(&p->tb_i, wi, 1, (uint64_t)p->N1 * split) != RNG_OK) return nullptr;
        }
        u64 nc[4] = {n, 0, 0, 0};
        p->ninv = Fr::from_canonical(nc).inverse();
        if (hipDeviceSynchronize() != hipSuccess) return nullptr;
        ctx->plans.emplace(n, std::move(np));
    }
    if (n > 4096 && p->scratch_elems < (uint64_t)n * batch) {
        if (p->scratch) hipFree(p->scratch);
        if (hipMalloc(&p->scratch, (uint64_t)n * batch * sizeof(Fr)) != hipSuccess) {
            p->scratch = nullptr;
            p->scratch_elems = 0;
            return nullptr;
        }
        p->scratch_elems = (uint64_t)n * batch;
    }
    return p;
}

// out-of-place for n > 4096 (result in out); in-place for small n.
static int ntt_dev_run(RngCtxImpl* ctx, Fr* data, Fr* out, uint32_t n, uint64_t batch,
                       bool inverse, hipStream_t stream = 0) {
    NttPlan* p = get_plan(ctx, n, batch);
    if (!p) return RNG_ERR_HIP;
    if (n <= 4096) {
        Fr* wst = inverse ? p->wst1_i : p->wst1_f;
        hipLaunchKernelGGL(k_ntt_small, dim3((uint32_t)batch), dim3(512), n * sizeof(Fr),
                           stream, data, wst, n, p->logn, p->ninv, inverse ? 1 : 0);
        HIP_CHECK(hipGetLastError());
        if (out != data)
            HIP_CHECK(hipMemcpyAsync(out, data, (uint64_t)n * batch * sizeof(Fr),
                                     hipMemcpyDeviceToDevice, stream));
        return RNG_OK;
    }
    Fr* wst1 = inverse ? p->wst1_i : p->wst1_f;
    Fr* wst2 = inverse ? p->wst2_i : p->wst2_f;
    Fr* ta = inverse ? p->ta_i : p->ta_f;
    Fr* tb = inverse ? p->tb_i : p->tb_f;
    uint32_t lds1 = 2 * p->N2 * sizeof(Fr);
    uint32_t lds2 = 2 * p->N1 * sizeof(Fr);
    EvtTimer et;
    et.mark(stream);
    hipLaunchKernelGGL(k_ntt_col, dim3((uint32_t)(p->N1 / 2 * batch)), dim3(512), lds1,
                       stream, data, wst2, ta, tb, p->N1, p->N2, p->logN2, p->split_log);
    HIP_CHECK(hipGetLastError());
    et.mark(stream);
    hipLaunchKernelGGL(k_ntt_row, dim3((uint32_t)(p->N2 / 2 * batch)), dim3(512), lds2,
                       stream, data, out, wst1, p->N1, p->N2, p->logN1, p->ninv,
                       inverse ? 1 : 0);
    HIP_CHECK(hipGetLastError());
    et.mark(stream);
    et.collect(tls_ntt_times, 2);
    return RNG_OK;
}

// ---------------- MSM ----------------

struct MsmScratch {
    uint32_t *keys_in = nullptr, *keys_out = nullptr, *vals_in = nullptr, *vals_out = nullptr;
    void* sort_temp = nullptr;
    size_t sort_temp_bytes = 0;
    uint8_t* head_flags = nullptr;
    uint32_t* heads = nullptr;
    uint32_t* heads_sorted = nullptr;
    uint32_t* lens = nullptr;
    uint32_t* lens_sorted = nullptr;
    uint32_t* head_count = nullptr;  // device u32
    void* select_temp = nullptr;
    size_t select_temp_bytes = 0;
    G1Jac* buckets = nullptr;
    G1Jac* partials = nullptr;
    G1Jac* window_sums = nullptr;
    G1Jac* result = nullptr;
    uint64_t cap_entries = 0;
    uint32_t cap_c = 0;

    ~MsmScratch() {
        for (void* b : {(void*)keys_in, (void*)keys_out, (void*)vals_in, (void*)vals_out,
                        sort_temp, (void*)head_flags, (void*)heads, (void*)heads_sorted,
                        (void*)lens, (void*)lens_sorted, (void*)head_count,
                        select_temp, (void*)buckets, (void*)partials, (void*)window_sums,
                        (void*)result})
            if (b) hipFree(b);
    }
};

static thread_local std::unique_ptr<MsmScratch> tls_msm_scratch;

static int msm_dev_run(const G1Aff* d_bases, const uint64_t* d_scalars, uint64_t n,
                       uint32_t c, G1Jac* h_result, hipStream_t stream = 0) {
    uint32_t W = (256 + c - 1) / c;
    uint64_t total = n * W;
    uint64_t nb = (1ull << (c - 1)) * W;  // total buckets
    uint64_t nchunks = ((1ull << (c - 1)) / MSM_CHUNK) * W;

    if (!tls_msm_scratch) tls_msm_scratch = std::make_unique<MsmScratch>();
    MsmScratch* s = tls_msm_scratch.get();
    if (s->cap_entries < total || s->cap_c != c) {
        tls_msm_scratch = std::make_unique<MsmScratch>();
        s = tls_msm_scratch.get();
        HIP_CHECK(hipMalloc(&s->keys_in, total * 4));
        HIP_CHECK(hipMalloc(&s->keys_out, total * 4));
        HIP_CHECK(hipMalloc(&s->vals_in, total * 4));
        HIP_CHECK(hipMalloc(&s->vals_out, total * 4));
        rocprim::radix_sort_pairs(nullptr, s->sort_temp_bytes, s->keys_in, s->keys_out,
                                  s->vals_in, s->vals_out, total, 0, 21, stream);
        HIP_CHECK(hipMalloc(&s->sort_temp, s->sort_temp_bytes));
        HIP_CHECK(hipMalloc(&s->head_flags, total));
        uint64_t max_heads_cap = (nb < total ? nb : total) + 1;
        HIP_CHECK(hipMalloc(&s->heads, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->heads_sorted, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->lens, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->lens_sorted, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->head_count, 4));
        {
            rocprim::counting_iterator<uint32_t> cit(0);
            (void)rocprim::select(nullptr, s->select_temp_bytes, cit, s->head_flags,
                                  s->heads, s->head_count, total, stream);
            HIP_CHECK(hipMalloc(&s->select_temp, s->select_temp_bytes));
        }
        HIP_CHECK(hipMalloc(&s->buckets, nb * sizeof(G1Jac)));
        HIP_CHECK(hipMalloc(&s->partials, 2 * nchunks * sizeof(G1Jac)));
        HIP_CHECK(hipMalloc(&s->window_sums, 32 * MSM_SUBB * sizeof(G1Jac)));
        HIP_CHECK(hipMalloc(&s->result, sizeof(G1Jac)));
        s->cap_entries = total;
        s->cap_c = c;
    }

    uint32_t tb = 256;
    EvtTimer et;
    et.mark(stream);
    hipLaunchKernelGGL(k_msm_digits, dim3((uint32_t)((n + tb - 1) / tb)), dim3(tb), 0,
                       stream, d_scalars, (uint32_t)n, c, W, s->keys_in, s->vals_in);
    HIP_CHECK(hipGetLastError());
    et.mark(stream);
    rocprim::radix_sort_pairs(s->sort_temp, s->sort_temp_bytes, s->keys_in, s->keys_out,
                              s->vals_in, s->vals_out, total, 0, 21, stream);
    HIP_CHECK(hipMemsetAsync(s->buckets, 0, nb * sizeof(G1Jac), stream));
    hipLaunchKernelGGL(k_msm_head_flags, dim3((uint32_t)((total + tb - 1) / tb)), dim3(tb),
                       0, stream, s->keys_out, (uint32_t)total, s->head_flags);
    HIP_CHECK(hipGetLastError());
    {
        rocprim::counting_iterator<uint32_t> cit(0);
        (void)rocprim::select(s->select_temp, s->select_temp_bytes, cit, s->head_flags,
                              s->heads, s->head_count, total, stream);
    }
    uint64_t max_heads = nb < total ? nb : total;
    hipLaunchKernelGGL(k_msm_seg_lengths, dim3((uint32_t)((max_heads + tb - 1) / tb)),
                       dim3(tb), 0, stream, s->keys_out, s->heads, s->head_count,
                       (uint32_t)total, s->lens);
    HIP_CHECK(hipGetLastError());
    // sort heads by segment length -> every wave walks near-equal segments
    // (rocprim needs a host-side item count: 4-byte DtoH + sync, ~10us)
    uint32_t hc = 0;
    HIP_CHECK(hipMemcpyAsync(&hc, s->head_count, 4, hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    if (hc > 0) {
        // descending: longest segments first so the tail waves pack short work
        rocprim::radix_sort_pairs_desc(s->sort_temp, s->sort_temp_bytes, s->lens,
                                       s->lens_sorted, s->heads, s->heads_sorted, hc, 0,
                                       25, stream);
        et.mark(stream);
        hipLaunchKernelGGL(k_msm_bucket_reduce, dim3((uint32_t)((hc + tb - 1) / tb)),
                           dim3(tb), 0, stream, s->keys_out, s->vals_out, s->heads_sorted,
                           s->lens_sorted, s->head_count, d_bases, s->buckets, c);
        HIP_CHECK(hipGetLastError());
    } else {
        et.mark(stream);
    }
    et.mark(stream);
    hipLaunchKernelGGL(k_msm_window_chunks, dim3((uint32_t)((nchunks + tb - 1) / tb)),
                       dim3(tb), 0, stream, s->buckets, c, W, s->partials);
    HIP_CHECK(hipGetLastError());
    et.mark(stream);
    hipLaunchKernelGGL(k_msm_window_combine, dim3(W * MSM_SUBB), dim3(64), 0, stream,
                       s->partials, c, s->window_sums);
    HIP_CHECK(hipGetLastError());
    et.mark(stream);
    // host-side fold: W*SUBB Jacobians (~1.5 KB); a single-lane dependent EC
    // chain is far faster on a host core than on one GPU lane
    G1Jac wsums[32 * MSM_SUBB];
    HIP_CHECK(hipMemcpyAsync(wsums, s->window_sums, W * MSM_SUBB * sizeof(G1Jac),
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    et.collect(tls_msm_times, 5);
    G1Jac ws[32];
    for (uint32_t w = 0; w < W; ++w) {
        G1Jac sum = wsums[w * MSM_SUBB];
        for (uint32_t i = 1; i < MSM_SUBB; ++i) sum = sum.add(wsums[w * MSM_SUBB + i]);
        ws[w] = sum;
    }
    G1Jac acc = ws[W - 1];
    for (int w = (int)W - 2; w >= 0; --w) {
        for (uint32_t k = 0; k < c; ++k) acc = acc.dbl();
        acc = acc.add(ws[w]);
    }
    *h_result = acc;
    return RNG_OK;
}

// Jacobian (host) -> 9-u64 affine record, using host-side field ops.
static void jac_to_affine_record(const G1Jac& j, uint64_t* out9) {
    if (j.Z.is_zero()) {
        memset(out9, 0, 9 * 8);
        out9[8] = 1;
        return;
    }
    Fq zinv = j.Z.inverse();
    Fq zinv2 = zinv.sqr();
    Fq x = j.X.mul(zinv2);
    Fq y = j.Y.mul(zinv2.mul(zinv));
    memcpy(out9, x.l, 32);
    memcpy(out9 + 4, y.l, 32);
    out9[8] = 0;
}

}  // namespace rng

// ---------------- C ABI ----------------

using namespace rng;

struct RngCtx {
    RngCtxImpl impl;
};

extern "C" {

const char* rng_version(void) { return "renegade_amd 0.1 (gfx950)"; }

int rng_gpu_available(void) { return gpu_ok() ? 1 : 0; }

int rng_set_device(int device) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    HIP_CHECK(hipSetDevice(device));
    return RNG_OK;
}

int rng_msm_last_times(double out_ms[5]) {
    for (int i = 0; i < 5; ++i) out_ms[i] = tls_msm_times[i];
    return 5;
}
int rng_ntt_last_times(double out_ms[2]) {
    for (int i = 0; i < 2; ++i) out_ms[i] = tls_ntt_times[i];
    return 2;
}

RngCtx* rng_prover_init(const uint8_t* srs_ptau, size_t len, uint64_t max_degree) {
    if (!srs_ptau || len < 12) return nullptr;
    auto ctx = std::make_unique<RngCtx>();
    RngCtxImpl* im = &ctx->impl;
    // --- parse ptau (semantics of srs.rs:63-214) ---
    size_t pos = 0;
    auto rd_u32 = [&](uint32_t* v) {
        if (pos + 4 > len) return false;
        memcpy(v, srs_ptau + pos, 4);
        pos += 4;
        return true;
    };
    auto rd_u64 = [&](uint64_t* v) {
        if (pos + 8 > len) return false;
        memcpy(v, srs_ptau + pos, 8);
        pos += 8;
        return true;
    };
    if (memcmp(srs_ptau, "ptau", 4) != 0) return nullptr;
    pos = 4;
    uint32_t version, nsections;
    if (!rd_u32(&version) || version != 1) return nullptr;
    if (!rd_u32(&nsections) || nsections != 11) return nullptr;
    uint32_t secnum;
    uint64_t secsize;
    if (!rd_u32(&secnum) || secnum != 1 || !rd_u64(&secsize)) return nullptr;
    size_t s1_end = pos + secsize;
    uint32_t n8;
    if (!rd_u32(&n8) || n8 != 32) return nullptr;
    static const u64 qmod[4] = FQ_MODULUS;
    if (pos + 32 > len || memcmp(srs_ptau + pos, qmod, 32) != 0) return nullptr;
    pos += 32;
    uint32_t power, cpower;
    if (!rd_u32(&power) || !rd_u32(&cpower)) return nullptr;
    if ((1ull << power) + 2 < max_degree) return nullptr;
    pos = s1_end;
    if (!rd_u32(&secnum) || secnum != 2 || !rd_u64(&secsize)) return nullptr;
    size_t s2_end = pos + secsize;
    uint64_t npoints = max_degree + 1;
    if (pos + npoints * 64 > len) return nullptr;
    im->srs_g1_host.resize(npoints * 8);
    memcpy(im->srs_g1_host.data(), srs_ptau + pos, npoints * 64);
    // curve-membership check (srs.rs:179) on a sample + endpoints; full check
    // deferred to the GPU parity tests to keep init latency low.
    {
        static const u64 bmont[4] = G1_B_MONT;
        Fq b;
        memcpy(b.l, bmont, 32);
        for (uint64_t i : {uint64_t(0), npoints / 2, npoints - 1}) {
            Fq x, y;
            memcpy(x.l, &im->srs_g1_host[i * 8], 32);
            memcpy(y.l, &im->srs_g1_host[i * 8 + 4], 32);
            if (!y.sqr().eq(x.sqr().mul(x).add(b))) return nullptr;
        }
    }
    pos = s2_end;
    if (!rd_u32(&secnum) || secnum != 3 || !rd_u64(&secsize)) return nullptr;
    if (pos + 2 * 128 > len) return nullptr;
    memcpy(im->h_g2, srs_ptau + pos, 128);
    memcpy(im->beta_h_g2, srs_ptau + pos + 128, 128);
    im->srs_count = npoints;
    // --- upload to GPU if present ---
    if (gpu_ok()) {
        if (hipMalloc(&im->srs_dev, npoints * 64) != hipSuccess) return nullptr;
        if (hipMemcpy(im->srs_dev, im->srs_g1_host.data(), npoints * 64,
                      hipMemcpyHostToDevice) != hipSuccess)
            return nullptr;
    }
    return ctx.release();
}

void rng_ctx_free(RngCtx* ctx) { delete ctx; }

const void* rng_srs_dev_bases(RngCtx* ctx, uint64_t* count) {
    if (count) *count = ctx->impl.srs_count;
    return ctx->impl.srs_dev;
}

void* rng_dbuf_alloc(size_t bytes) {
    void* p = nullptr;
    if (hipMalloc(&p, bytes) != hipSuccess) return nullptr;
    return p;
}
void rng_dbuf_free(void* p) {
    if (p) hipFree(p);
}
int rng_dbuf_upload(void* d, const void* h, size_t bytes) {
    HIP_CHECK(hipMemcpy(d, h, bytes, hipMemcpyHostToDevice));
    return RNG_OK;
}
int rng_dbuf_download(const void* d, void* h, size_t bytes) {
    HIP_CHECK(hipMemcpy(h, (void*)d, bytes, hipMemcpyDeviceToHost));
    return RNG_OK;
}
int rng_device_sync(void) {
    HIP_CHECK(hipDeviceSynchronize());
    return RNG_OK;
}

int rng_ntt_fr_dev(RngCtx* ctx, void* dev_data, uint64_t n, uint64_t batch, int inverse) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!ctx || !dev_data || n < 2 || (n & (n - 1)) || n > (1ull << 26))
        return RNG_ERR_BAD_ARG;
    Fr* data = (Fr*)dev_data;
    NttPlan* p = get_plan(&ctx->impl, (uint32_t)n, batch);
    if (!p) return RNG_ERR_HIP;
    if (n <= 4096) return ntt_dev_run(&ctx->impl, data, data, (uint32_t)n, batch, inverse);
    int rc = ntt_dev_run(&ctx->impl, data, p->scratch, (uint32_t)n, batch, inverse);
    if (rc != RNG_OK) return rc;
    HIP_CHECK(hipMemcpy(data, p->scratch, n * batch * sizeof(Fr), hipMemcpyDeviceToDevice));
    return RNG_OK;
}

int rng_ntt_fr_dev_oop(RngCtx* ctx, void* dev_in, void* dev_out, uint64_t n,
                       uint64_t batch, int inverse) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!ctx || !dev_in || !dev_out || n < 2 || (n & (n - 1)) || n > (1ull << 26))
        return RNG_ERR_BAD_ARG;
    return ntt_dev_run(&ctx->impl, (Fr*)dev_in, (Fr*)dev_out, (uint32_t)n, batch, inverse);
}

int rng_ntt_fr(RngCtx* ctx, uint64_t* data, uint64_t n, uint64_t batch, int inverse) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!ctx || !data || n < 2 || (n & (n - 1)) || n > (1ull << 26)) return RNG_ERR_BAD_ARG;
    size_t bytes = n * batch * sizeof(Fr);
    Fr* d = nullptr;
    HIP_CHECK(hipMalloc(&d, bytes));
    HIP_CHECK(hipMemcpy(d, data, bytes, hipMemcpyHostToDevice));
    NttPlan* p = get_plan(&ctx->impl, (uint32_t)n, batch);
    if (!p) {
        hipFree(d);
        return RNG_ERR_HIP;
    }
    Fr* out = (n <= 4096) ? d : p->scratch;
    int rc = ntt_dev_run(&ctx->impl, d, out, (uint32_t)n, batch, inverse);
    if (rc == RNG_OK) {
        if (hipMemcpy(data, out, bytes, hipMemcpyDeviceToHost) != hipSuccess)
            rc = RNG_ERR_HIP;
    }
    hipFree(d);
    return rc;
}

int rng_msm_g1_dev(RngCtx* ctx, const void* dev_bases, const void* dev_scalars,
                   uint64_t n, uint64_t* out9, int window_c) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!dev_bases || !dev_scalars || !out9 || n == 0) return RNG_ERR_BAD_ARG;
    uint32_t c = window_c > 0 ? (uint32_t)window_c : 16;
    if (c < 8 || c > 16) return RNG_ERR_BAD_ARG;
    G1Jac res;
    int rc = msm_dev_run((const G1Aff*)dev_bases, (const uint64_t*)dev_scalars, n, c, &res);
    if (rc != RNG_OK) return rc;
    jac_to_affine_record(res, out9);
    return RNG_OK;
}

int rng_msm_g1(RngCtx* ctx, const uint64_t* bases, const uint64_t* scalars, uint64_t n,
               uint64_t* out9, int window_c) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!bases || !scalars || !out9 || n == 0) return RNG_ERR_BAD_ARG;
    void *db = nullptr, *ds = nullptr;
    HIP_CHECK(hipMalloc(&db, n * 64));
    if (hipMalloc(&ds, n * 32) != hipSuccess) {
        hipFree(db);
        return RNG_ERR_HIP;
    }
    int rc = RNG_OK;
    if (hipMemcpy(db, bases, n * 64, hipMemcpyHostToDevice) != hipSuccess ||
        hipMemcpy(ds, scalars, n * 32, hipMemcpyHostToDevice) != hipSuccess)
        rc = RNG_ERR_HIP;
    if (rc == RNG_OK) rc = rng_msm_g1_dev(ctx, db, ds, n, out9, window_c);
    hipFree(db);
    hipFree(ds);
    return rc;
}

// ---- circuit construction (arithmetization front-end) ----
// The tables handle wraps rng::CircuitTables; getters copy flat arrays out
// so tests (and the oracle prover) consume identical inputs.

void* rng_testcirc_build(uint64_t seed, uint64_t scale) {
    try {
        PlonkCircuit cs;
        build_mixed_circuit(cs, seed, scale);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_testcirc_build: %s\n", why.c_str());
            return nullptr;
        }
        auto* t = new CircuitTables(cs.finalize());
        return t;
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_testcirc_build: %s\n", e.what());
        return nullptr;
    }
}

uint64_t rng_circ_n(void* t) { return static_cast<CircuitTables*>(t)->n; }
uint64_t rng_circ_npub(void* t) { return static_cast<CircuitTables*>(t)->num_public; }

// selectors: 13*n*4 u64; sigma: 5*n u64; wires: 5*n*4 u64; pubs: npub*4 u64
void rng_circ_get(void* t_, uint64_t* selectors, uint64_t* sigma, uint64_t* wires,
                  uint64_t* pubs) {
    auto* t = static_cast<CircuitTables*>(t_);
    memcpy(selectors, t->selectors.data(), t->selectors.size() * sizeof(Fr));
    memcpy(sigma, t->sigma.data(), t->sigma.size() * 8);
    memcpy(wires, t->wires.data(), t->wires.size() * sizeof(Fr));
    if (!t->public_inputs.empty())
        memcpy(pubs, t->public_inputs.data(), t->public_inputs.size() * sizeof(Fr));
}

void rng_circ_free(void* t) { delete static_cast<CircuitTables*>(t); }

// Plonk-layer entry points land with the plonk milestone (DESIGN.md roadmap);
// fail loudly rather than silently succeed.
RngProvingKey* rng_preprocess(RngCtx*, const RngCircuitDesc*) { return nullptr; }
void rng_pk_free(RngProvingKey*) {}
int rng_prove(RngCtx*, const RngProvingKey*, const uint64_t*, const uint64_t*, uint64_t,
              uint64_t*, uint64_t*) {
    return RNG_ERR_BAD_ARG;
}
int rng_verify(RngCtx*, const RngProvingKey*, const uint64_t*, const uint64_t*) {
    return RNG_ERR_BAD_ARG;
}
int rng_link_proofs(RngCtx*, const RngProvingKey*, const uint64_t*, const uint64_t*,
                    uint64_t, uint64_t, uint64_t*) {
    return RNG_ERR_BAD_ARG;
}

}  // extern "C"
