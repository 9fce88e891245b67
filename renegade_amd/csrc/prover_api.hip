// PRODUCT PATH — C-ABI implementation (include/rng_prover.h).
//
// Host orchestration for the MI355X prover backend: SRS load (ptau
// semantics of crates/circuits/circuit-types/src/primitives/srs.rs:63-214),
// NTT plans, MSM pipeline.  Single translation unit: includes the kernel
// files directly.
#include <array>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdio>
#include <functional>
#include <cstdlib>
#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <thread>
#include <vector>

#include "../../include/rng_prover.h"
#include "ntt_kernels.hip"
#include "msm_kernels.hip"
#include "plonk_kernels.hip"
#include "plonk_circuit.hpp"
#include "plonk_host.hpp"
#include "test_circuits.hpp"
#include "circuits_core.hpp"
#include "pairing_impl.hpp"

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

// every prover-path op runs on the calling thread's implicit stream so
// concurrent rng_prove calls from a thread pool overlap on the GPU
#define RNG_STREAM hipStreamPerThread

// Teardown guard: main-thread TLS / static / Python-GC destructors can run
// AFTER the HIP runtime's own atexit teardown, and a hipFree then faults
// inside the dead runtime (r01 bench exited rc=139 this way).  We register
// our atexit handler AFTER the first successful HIP call, so (LIFO order) it
// runs BEFORE the runtime's teardown and flips this flag; every destructor
// hipFree is gated on it.  Device memory skipped this way is reclaimed by
// process exit anyway.
static std::atomic<bool> g_hip_alive{true};
static bool hip_alive() { return g_hip_alive.load(std::memory_order_relaxed); }
static void hip_free_guarded(void* p) {
    if (p && hip_alive()) hipFree(p);
}

static bool gpu_ok() {
    static int cached = -1;
    if (cached < 0) {
        int n = 0;
        cached = (hipGetDeviceCount(&n) == hipSuccess && n > 0) ? 1 : 0;
        if (cached == 1)
            std::atexit([] { g_hip_alive.store(false, std::memory_order_relaxed); });
    }
    return cached == 1;
}

static thread_local double tls_msm_times[5] = {0, 0, 0, 0, 0};
static thread_local double tls_ntt_times[2] = {0, 0};

// Persistent host worker pool: parallelizes the per-proof host phases of
// cohort proving (grand product, evaluations, linearization) and the
// per-poly window folds of large fused MSM batches.  One job at a time;
// concurrent callers queue.  Never destroyed (leaked) so there is no
// thread-join ordering problem at process exit.
class HostPool {
    // Each job is a heap object shared by every thread that touches it: a
    // straggler from job A holds A's shared_ptr, sees A's exhausted `next`
    // counter and exits — it can never execute job B's indices with A's
    // (destroyed) callable.
    struct Job {
        std::function<void(uint32_t)> fn;
        uint32_t count;
        std::atomic<uint32_t> next{0}, done{0};
    };

  public:
    static HostPool& inst() {
        static HostPool* p = new HostPool();  // leaked: no join-at-exit problem
        return *p;
    }

    void parallel_for(uint32_t count, const std::function<void(uint32_t)>& fn) {
        if (count == 0) return;
        if (count == 1 || workers_.empty()) {
            for (uint32_t i = 0; i < count; ++i) fn(i);
            return;
        }
        std::unique_lock<std::mutex> job_lk(job_mu_);  // one job at a time
        auto job = std::make_shared<Job>();
        job->fn = fn;
        job->count = count;
        {
            std::lock_guard<std::mutex> lk(mu_);
            job_ = job;
            ++generation_;
        }
        cv_.notify_all();
        run_job(*job);  // caller participates
        {
            std::unique_lock<std::mutex> lk(mu_);
            done_cv_.wait(lk, [&] {
                return job->done.load(std::memory_order_acquire) >= count;
            });
            job_.reset();
        }
    }

  private:
    HostPool() {
        unsigned n = std::thread::hardware_concurrency();
        if (n == 0) n = 4;
        if (n > 32) n = 32;
        for (unsigned t = 0; t + 1 < n; ++t)
            workers_.emplace_back([this] { worker_loop(); });
    }
    void run_job(Job& job) {
        for (;;) {
            uint32_t i = job.next.fetch_add(1, std::memory_order_relaxed);
            if (i >= job.count) break;
            job.fn(i);
            if (job.done.fetch_add(1, std::memory_order_acq_rel) + 1 >= job.count) {
                std::lock_guard<std::mutex> lk(mu_);
                done_cv_.notify_all();
            }
        }
    }
    void worker_loop() {
        uint64_t seen = 0;
        for (;;) {
            std::shared_ptr<Job> job;
            {
                std::unique_lock<std::mutex> lk(mu_);
                cv_.wait(lk, [&] { return generation_ != seen || stopping_; });
                if (stopping_) return;
                seen = generation_;
                job = job_;
            }
            if (job) run_job(*job);
        }
    }
    std::mutex job_mu_;
    std::mutex mu_;
    std::condition_variable cv_, done_cv_;
    std::vector<std::thread> workers_;
    std::shared_ptr<Job> job_;
    uint64_t generation_ = 0;
    bool stopping_ = false;

  public:
    // Join every worker (profilers' finalizers can stall on foreign live
    // threads at process exit; bench/debug scripts call rng_shutdown_pool
    // before exiting under rocprof).  parallel_for falls back to serial
    // execution afterwards.
    void stop() {
        std::unique_lock<std::mutex> job_lk(job_mu_);
        {
            std::lock_guard<std::mutex> lk(mu_);
            if (stopping_) return;
            stopping_ = true;
        }
        cv_.notify_all();
        for (auto& t : workers_) t.join();
        workers_.clear();
    }
};

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
    // coset tables (built on demand): gpow g^j, gpow_inv g^-j, xpow g*w^j
    Fr *gpow = nullptr, *gpow_inv = nullptr, *xpow = nullptr;
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
    void* srs_glv_dev = nullptr;  // [P, phi(P)] interleaved for GLV
    uint64_t h_g2[16];       // h: x.c0,x.c1,y.c0,y.c1 Montgomery
    uint64_t beta_h_g2[16];
    std::map<uint32_t, std::unique_ptr<NttPlan>> plans;
    std::mutex mu;

    ~RngCtxImpl() {
        for (auto& kv : plans) {
            NttPlan* p = kv.second.get();
            for (Fr* b : {p->wst1_f, p->wst2_f, p->ta_f, p->tb_f, p->wst1_i,
                          p->wst2_i, p->ta_i, p->tb_i, p->scratch, p->gpow,
                          p->gpow_inv, p->xpow})
                hip_free_guarded(b);
        }
        hip_free_guarded(srs_dev);
        hip_free_guarded(srs_glv_dev);
    }
};

static int build_pow_table(Fr** out, const Fr& base, uint64_t step, uint64_t count) {
    HIP_CHECK(hipMalloc(out, count * sizeof(Fr)));
    uint32_t blocks = (uint32_t)((count + 255) / 256);
    hipLaunchKernelGGL(k_pow_table, dim3(blocks), dim3(256), 0, 0, *out, base, step, count);
    HIP_CHECK(hipGetLastError());
    return RNG_OK;
}

// single-workgroup NTT threshold: one WG per column is latency-bound (12
// serial stages), so route n >= 2048 through the two-pass col/row path
// (64-wide grids) — measured 3x faster at n=4096 (the proof domain).
static constexpr uint32_t NTT_SMALL_MAX = 1024;

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
        if (n <= NTT_SMALL_MAX) {
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
    if (n > NTT_SMALL_MAX && p->scratch_elems < (uint64_t)n * batch) {
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

// out-of-place for n > NTT_SMALL_MAX (result in out); in-place for small n.
static int ntt_dev_run(RngCtxImpl* ctx, Fr* data, Fr* out, uint32_t n, uint64_t batch,
                       bool inverse, hipStream_t stream = 0) {
    // batch=0 to get_plan: tables only — ntt_dev_run never touches the
    // plan's public-API scratch (rng_ntt_fr_dev sizes it itself)
    NttPlan* p = get_plan(ctx, n, 0);
    if (!p) return RNG_ERR_HIP;
    if (n <= NTT_SMALL_MAX) {
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
    // pair width per pass: 2 (64-byte-coalesced global access) while the
    // pair fits 64 KiB LDS (>=2 blocks/CU); above that, singles — at 2^22
    // the 128 KiB pair form pinned the DFT at 1 block/CU and the dependent
    // butterfly muls ran latency-exposed
    static uint32_t pw_lds_cap = [] {
        const char* e = getenv("RNG_NTT_PAIR_LDS");  // A/B knob, bytes
        return e ? (uint32_t)atoi(e) : 65536u;
    }();
    const uint32_t pw1 = (2u * p->N2 * sizeof(Fr) <= pw_lds_cap) ? 2 : 1;
    const uint32_t pw2 = (2u * p->N1 * sizeof(Fr) <= pw_lds_cap) ? 2 : 1;
    uint32_t lds1 = pw1 * p->N2 * sizeof(Fr);
    uint32_t lds2 = pw2 * p->N1 * sizeof(Fr);
    EvtTimer et;
    et.mark(stream);
    if (pw1 == 2)
        hipLaunchKernelGGL(k_ntt_col<2>, dim3((uint32_t)(p->N1 / 2 * batch)), dim3(512),
                           lds1, stream, data, wst2, ta, tb, p->N1, p->N2, p->logN2,
                           p->split_log);
    else
        hipLaunchKernelGGL(k_ntt_col<1>, dim3((uint32_t)(p->N1 * batch)), dim3(512),
                           lds1, stream, data, wst2, ta, tb, p->N1, p->N2, p->logN2,
                           p->split_log);
    HIP_CHECK(hipGetLastError());
    et.mark(stream);
    if (pw2 == 2)
        hipLaunchKernelGGL(k_ntt_row<2>, dim3((uint32_t)(p->N2 / 2 * batch)), dim3(512),
                           lds2, stream, data, out, wst1, p->N1, p->N2, p->logN1,
                           p->ninv, inverse ? 1 : 0);
    else
        hipLaunchKernelGGL(k_ntt_row<1>, dim3((uint32_t)(p->N2 * batch)), dim3(512),
                           lds2, stream, data, out, wst1, p->N1, p->N2, p->logN1,
                           p->ninv, inverse ? 1 : 0);
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
    uint32_t* lens = nullptr;
    uint32_t* nsub = nullptr;
    uint32_t* sub_off = nullptr;
    uint32_t* sub_start = nullptr;
    uint32_t* sub_len = nullptr;
    uint32_t* sub_len_sorted = nullptr;
    uint32_t* sub_order = nullptr;
    uint32_t* nsub_sorted = nullptr;
    uint32_t* head_order = nullptr;
    G1Jac* partials2 = nullptr;
    uint32_t* head_count = nullptr;  // device u32
    void* select_temp = nullptr;
    size_t select_temp_bytes = 0;
    void* scan_temp = nullptr;
    size_t scan_temp_bytes = 0;
    // binned (counting-scatter) pipeline
    uint32_t* counts = nullptr;   // nb
    uint32_t* offsets = nullptr;  // nb
    uint32_t* cursor = nullptr;   // nb
    uint8_t* bflags = nullptr;    // nb
    G1Jac* buckets = nullptr;
    G1Jac* partials = nullptr;
    G1Jac* window_sums = nullptr;
    uint64_t window_sums_cap = 0;  // entries (G*subb grows with fused B)
    G1Jac* result = nullptr;
    uint64_t* glv = nullptr;   // per-scalar (k1, k2) magnitudes + signs
    G1Aff* phi = nullptr;      // phi(bases) scratch for non-SRS base arrays
    uint64_t cap_entries = 0;
    uint64_t cap_nb = 0;
    uint64_t cap_nchunks = 0;

    ~MsmScratch() {
        for (void* b : {(void*)keys_in, (void*)keys_out, (void*)vals_in, (void*)vals_out,
                        sort_temp, (void*)head_flags, (void*)heads, (void*)lens,
                        (void*)nsub, (void*)sub_off, (void*)sub_start, (void*)sub_len,
                        (void*)sub_len_sorted, (void*)sub_order,
                        (void*)nsub_sorted, (void*)head_order,
                        (void*)partials2, (void*)head_count, select_temp, scan_temp,
                        (void*)buckets, (void*)partials, (void*)window_sums,
                        (void*)result, (void*)glv, (void*)phi, (void*)counts,
                        (void*)offsets, (void*)cursor, (void*)bflags})
            hip_free_guarded(b);
    }
};

static thread_local std::unique_ptr<MsmScratch> tls_msm_scratch;

// B polynomials sharing one base array -> one fused pipeline; results[B].
// GLV endomorphism path (RNG_MSM_GLV=1): bit-exact (parity-green) but OFF
// by default — measured on MI355X the halved aggregation does not pay for
// the decompose + doubled digit work at proof sizes (335 vs 387 proofs/s)
// or at 2^20 (7.4 vs 6.3 ms); see DESIGN.md §4.1.
static inline bool msm_glv_enabled() {
    static int v = [] {
        const char* e = getenv("RNG_MSM_GLV");
        return e ? atoi(e) : 0;
    }();
    return v != 0;
}

static int msm_dev_run(const G1Aff* d_bases, const uint64_t* d_scalars, uint64_t n,
                       uint32_t c, G1Jac* h_result, uint32_t B = 1,
                       hipStream_t stream = RNG_STREAM,
                       const G1Aff* d_glv_bases = nullptr) {
    const bool glv = msm_glv_enabled();
    // GLV halves scalar width: two 127-bit halves share each window's buckets
    uint32_t W = glv ? (128 + c - 1) / c : (256 + c - 1) / c;
    uint32_t G = B * W;  // key groups
    uint64_t per_scalar_entries = glv ? 2 * W : W;
    uint64_t total = n * B * per_scalar_entries;
    uint64_t nb = (1ull << (c - 1)) * G;  // total buckets
    // chunk size adapts to keep the window-sum stages wide: small MSMs (few
    // windows x small bucket counts) get chunk=2 (4x the lanes, 4x shorter
    // latency chains), the 2^20 leg keeps chunk=16.
    uint32_t chunk_sz = MSM_CHUNK;
    {
        uint64_t nbw = (1ull << (c - 1)) * G;
        // target >=128k lanes: the per-chunk suffix walk is a serial EC
        // chain, so the chunk kernel is latency-bound until the chip is
        // several waves deep per SIMD (r02: chunk 16 -> 4 at 2^20 cut
        // window_chunks ~3x)
        while (chunk_sz > 2 && nbw / chunk_sz < 131072) chunk_sz >>= 1;
        static int chunk_env = [] {
            const char* e = getenv("RNG_MSM_CHUNK");
            return e ? atoi(e) : 0;
        }();
        if (chunk_env >= 2 && chunk_env <= 16) chunk_sz = (uint32_t)chunk_env;
        if ((1u << (c - 1)) < chunk_sz) chunk_sz = 1u << (c - 1);
    }
    uint64_t nchunks = ((1ull << (c - 1)) / chunk_sz) * G;

    if (!tls_msm_scratch) tls_msm_scratch = std::make_unique<MsmScratch>();
    MsmScratch* s = tls_msm_scratch.get();
    if (s->cap_entries < total || s->cap_nb < nb || s->cap_nchunks < nchunks) {
        // size with 1.5x slack so nearby problem sizes (n, n+2, n+3 commits,
        // different auto window sizes) never thrash device allocations
        uint64_t cap_total = total + total / 2 + 64;
        uint64_t cap_nb2 = nb + nb / 2 + 64;
        uint64_t cap_nch = nchunks + nchunks / 2 + 64;
        tls_msm_scratch = std::make_unique<MsmScratch>();
        s = tls_msm_scratch.get();
        HIP_CHECK(hipMalloc(&s->keys_in, cap_total * 4));
        HIP_CHECK(hipMalloc(&s->keys_out, cap_total * 4));
        HIP_CHECK(hipMalloc(&s->vals_in, cap_total * 4));
        HIP_CHECK(hipMalloc(&s->vals_out, cap_total * 4));
        rocprim::radix_sort_pairs(nullptr, s->sort_temp_bytes, s->keys_in, s->keys_out,
                                  s->vals_in, s->vals_out, cap_total, 0, 27, stream);
        HIP_CHECK(hipMalloc(&s->sort_temp, s->sort_temp_bytes));
        HIP_CHECK(hipMalloc(&s->head_flags, cap_total));
        uint64_t max_heads_cap = (cap_nb2 < cap_total ? cap_nb2 : cap_total) + 1;
        uint64_t max_subs_cap = max_heads_cap + cap_total / 8 + 1;
        HIP_CHECK(hipMalloc(&s->heads, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->lens, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->nsub, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->sub_off, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->sub_start, max_subs_cap * 4));
        HIP_CHECK(hipMalloc(&s->sub_len, max_subs_cap * 4));
        HIP_CHECK(hipMalloc(&s->sub_len_sorted, max_subs_cap * 4));
        HIP_CHECK(hipMalloc(&s->sub_order, max_subs_cap * 4));
        HIP_CHECK(hipMalloc(&s->nsub_sorted, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->head_order, max_heads_cap * 4));
        HIP_CHECK(hipMalloc(&s->partials2, max_subs_cap * sizeof(G1Jac)));
        HIP_CHECK(hipMalloc(&s->head_count, 4));
        {
            rocprim::counting_iterator<uint32_t> cit(0);
            uint64_t sel_cap = cap_total > cap_nb2 ? cap_total : cap_nb2;
            (void)rocprim::select(nullptr, s->select_temp_bytes, cit, s->head_flags,
                                  s->heads, s->head_count, sel_cap, stream);
            HIP_CHECK(hipMalloc(&s->select_temp, s->select_temp_bytes));
            uint64_t scan_cap = max_heads_cap > cap_nb2 ? max_heads_cap : cap_nb2;
            (void)rocprim::exclusive_scan(nullptr, s->scan_temp_bytes, s->nsub,
                                          s->sub_off, 0u, scan_cap,
                                          rocprim::plus<uint32_t>(), stream);
            HIP_CHECK(hipMalloc(&s->scan_temp, s->scan_temp_bytes));
        }
        HIP_CHECK(hipMalloc(&s->counts, cap_nb2 * 4));
        HIP_CHECK(hipMalloc(&s->offsets, cap_nb2 * 4));
        HIP_CHECK(hipMalloc(&s->cursor, cap_nb2 * 4));
        HIP_CHECK(hipMalloc(&s->bflags, cap_nb2));
        HIP_CHECK(hipMalloc(&s->buckets, cap_nb2 * sizeof(G1Jac)));
        HIP_CHECK(hipMalloc(&s->partials, 2 * cap_nch * sizeof(G1Jac)));
        HIP_CHECK(hipMalloc(&s->window_sums, 512 * MSM_SUBB * sizeof(G1Jac)));
        s->window_sums_cap = 512 * MSM_SUBB;
        HIP_CHECK(hipMalloc(&s->result, sizeof(G1Jac)));
        // glv: 32 B per scalar = 16*total/W bytes <= 2*total (W >= 8)
        HIP_CHECK(hipMalloc(&s->glv, 2 * cap_total));
        // phi: 64 B per base = 32*total/(W*B) bytes <= 4*total (W >= 8, B >= 1)
        HIP_CHECK(hipMalloc(&s->phi, 4 * cap_total));
        s->cap_entries = cap_total;
        s->cap_nb = cap_nb2;
        s->cap_nchunks = cap_nch;
    }

    // binned (counting-scatter) pipeline, RNG_MSM_BINNED=1 to enable.
    // Measured SLOWER than the rocPRIM onesweep sort on MI355X (2^20 MSM:
    // digits+hist 0.63 ms with hot-counter atomics + scatter 1.60 ms of
    // uncoalesced atomic-cursor writes vs 0.04 + 0.86 for digits+sort;
    // cohort headline 1026 vs 1071 proofs/s) — kept for A/B re-checks.
    static int binned_env = [] {
        const char* e = getenv("RNG_MSM_BINNED");
        return e ? atoi(e) : 0;
    }();
    const bool binned = binned_env && !glv;

    uint32_t tb = 256;
    EvtTimer et;
    et.mark(stream);
    if (binned) {
        HIP_CHECK(hipMemsetAsync(s->counts, 0, nb * 4, stream));
        hipLaunchKernelGGL(k_msm_digits_hist, dim3((uint32_t)((n * B + tb - 1) / tb)),
                           dim3(tb), 0, stream, d_scalars, (uint32_t)n, c, W, B,
                           s->keys_in, s->vals_in, s->counts);
        HIP_CHECK(hipGetLastError());
        et.mark(stream);
        (void)rocprim::exclusive_scan(s->scan_temp, s->scan_temp_bytes, s->counts,
                                      s->offsets, 0u, nb, rocprim::plus<uint32_t>(),
                                      stream);
        HIP_CHECK(hipMemcpyAsync(s->cursor, s->offsets, nb * 4,
                                 hipMemcpyDeviceToDevice, stream));
        hipLaunchKernelGGL(k_msm_scatter, dim3((uint32_t)((total + tb - 1) / tb)),
                           dim3(tb), 0, stream, s->keys_in, s->vals_in, total,
                           s->cursor, s->vals_out);
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipMemsetAsync(s->buckets, 0, nb * sizeof(G1Jac), stream));
        hipLaunchKernelGGL(k_nonzero_flags, dim3((uint32_t)((nb + tb - 1) / tb)),
                           dim3(tb), 0, stream, s->counts, (uint32_t)nb, s->bflags);
        HIP_CHECK(hipGetLastError());
        {
            // keys_out holds the compacted non-empty bucket ids
            rocprim::counting_iterator<uint32_t> cit(0);
            (void)rocprim::select(s->select_temp, s->select_temp_bytes, cit, s->bflags,
                                  s->keys_out, s->head_count, nb, stream);
        }
        uint32_t seg_cap = msm_seg_cap(total);
        hipLaunchKernelGGL(k_msm_seg_from_counts, dim3((uint32_t)((nb + tb - 1) / tb)),
                           dim3(tb), 0, stream, s->keys_out, s->offsets, s->counts,
                           s->head_count, s->heads, s->lens, s->nsub, seg_cap);
        HIP_CHECK(hipGetLastError());
        uint32_t hc = 0;
        HIP_CHECK(hipMemcpyAsync(&hc, s->head_count, 4, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        if (hc > 0) {
            (void)rocprim::exclusive_scan(s->scan_temp, s->scan_temp_bytes, s->nsub,
                                          s->sub_off, 0u, hc, rocprim::plus<uint32_t>(),
                                          stream);
            uint32_t last_off = 0, last_n = 0;
            HIP_CHECK(hipMemcpyAsync(&last_off, s->sub_off + hc - 1, 4,
                                     hipMemcpyDeviceToHost, stream));
            HIP_CHECK(hipMemcpyAsync(&last_n, s->nsub + hc - 1, 4,
                                     hipMemcpyDeviceToHost, stream));
            HIP_CHECK(hipStreamSynchronize(stream));
            uint32_t sub_total = last_off + last_n;
            hipLaunchKernelGGL(k_msm_make_subs, dim3((hc + tb - 1) / tb), dim3(tb), 0,
                               stream, s->heads, s->lens, s->sub_off, s->head_count,
                               s->sub_start, s->sub_len, seg_cap);
            HIP_CHECK(hipGetLastError());
            {
                rocprim::counting_iterator<uint32_t> cit(0);
                rocprim::radix_sort_pairs_desc(s->sort_temp, s->sort_temp_bytes,
                                               s->sub_len, s->sub_len_sorted, cit,
                                               s->sub_order, sub_total, 0, 8, stream);
            }
            et.mark(stream);
            hipLaunchKernelGGL(k_msm_bucket_reduce, dim3((sub_total + tb - 1) / tb),
                               dim3(tb), 0, stream, s->vals_out, s->sub_start,
                               s->sub_len, s->sub_order, sub_total, d_bases,
                               s->partials2);
            HIP_CHECK(hipGetLastError());
            {
                rocprim::counting_iterator<uint32_t> cit(0);
                rocprim::radix_sort_pairs_desc(s->sort_temp, s->sort_temp_bytes,
                                               s->nsub, s->nsub_sorted, cit,
                                               s->head_order, hc, 0, 16, stream);
            }
            hipLaunchKernelGGL(k_msm_seg_merge2, dim3((hc + tb - 1) / tb), dim3(tb), 0,
                               stream, s->keys_out, s->sub_off, s->nsub, s->head_order,
                               hc, s->partials2, s->buckets);
            HIP_CHECK(hipGetLastError());
        } else {
            et.mark(stream);
        }
    } else {  // radix-sort path (GLV, or RNG_MSM_BINNED=0)
    if (glv) {
        if (!d_glv_bases) {  // non-SRS bases: interleave into scratch
            hipLaunchKernelGGL(k_bases_interleave, dim3((uint32_t)((n + tb - 1) / tb)),
                               dim3(tb), 0, stream, d_bases, s->phi, (uint32_t)n);
            HIP_CHECK(hipGetLastError());
            d_glv_bases = s->phi;
        }
        d_bases = d_glv_bases;  // bucket gathers index the interleaved array
        hipLaunchKernelGGL(k_glv_decompose, dim3((uint32_t)((n * B + tb - 1) / tb)),
                           dim3(tb), 0, stream, d_scalars, (uint32_t)(n * B), s->glv);
        HIP_CHECK(hipGetLastError());
        hipLaunchKernelGGL(k_msm_digits_glv, dim3((uint32_t)((n * B + tb - 1) / tb)),
                           dim3(tb), 0, stream, s->glv, (uint32_t)n, c, W, B,
                           s->keys_in, s->vals_in);
        HIP_CHECK(hipGetLastError());
    } else {
        hipLaunchKernelGGL(k_msm_digits, dim3((uint32_t)((n * B + tb - 1) / tb)),
                           dim3(tb), 0, stream, d_scalars, (uint32_t)n, c, W, B,
                           s->keys_in, s->vals_in);
        HIP_CHECK(hipGetLastError());
    }
    et.mark(stream);
    // sort only the live key bits: 16 magnitude bits + enough for G groups
    // + the sentinel (G << 16); 3 radix passes instead of 4
    uint32_t gb = 1;
    while ((1u << gb) <= G) ++gb;
    rocprim::radix_sort_pairs(s->sort_temp, s->sort_temp_bytes, s->keys_in, s->keys_out,
                              s->vals_in, s->vals_out, total, 0, 16 + gb, stream);
    HIP_CHECK(hipMemsetAsync(s->buckets, 0, nb * sizeof(G1Jac), stream));
    hipLaunchKernelGGL(k_msm_head_flags, dim3((uint32_t)((total + tb - 1) / tb)), dim3(tb),
                       0, stream, s->keys_out, (uint32_t)total, msm_sentinel(G),
                       s->head_flags);
    HIP_CHECK(hipGetLastError());
    {
        rocprim::counting_iterator<uint32_t> cit(0);
        (void)rocprim::select(s->select_temp, s->select_temp_bytes, cit, s->head_flags,
                              s->heads, s->head_count, total, stream);
    }
    uint64_t max_heads = nb < total ? nb : total;
    uint32_t seg_cap = msm_seg_cap(total);
    hipLaunchKernelGGL(k_msm_seg_lengths, dim3((uint32_t)((max_heads + tb - 1) / tb)),
                       dim3(tb), 0, stream, s->keys_out, s->heads, s->head_count,
                       (uint32_t)total, s->lens, s->nsub, seg_cap);
    HIP_CHECK(hipGetLastError());
    uint32_t hc = 0;
    HIP_CHECK(hipMemcpyAsync(&hc, s->head_count, 4, hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    if (hc > 0) {
        // sub-segment split (caps any lane's walk at MSM_MAX_SEG entries)
        (void)rocprim::exclusive_scan(s->scan_temp, s->scan_temp_bytes, s->nsub,
                                      s->sub_off, 0u, hc, rocprim::plus<uint32_t>(),
                                      stream);
        uint32_t last_off = 0, last_n = 0;
        HIP_CHECK(hipMemcpyAsync(&last_off, s->sub_off + hc - 1, 4,
                                 hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipMemcpyAsync(&last_n, s->nsub + hc - 1, 4, hipMemcpyDeviceToHost,
                                 stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        uint32_t sub_total = last_off + last_n;
        hipLaunchKernelGGL(k_msm_make_subs, dim3((hc + tb - 1) / tb), dim3(tb), 0,
                           stream, s->heads, s->lens, s->sub_off, s->head_count,
                           s->sub_start, s->sub_len, seg_cap);
        HIP_CHECK(hipGetLastError());
        {
            // order subs by length desc so each wave's lanes walk equal work
            rocprim::counting_iterator<uint32_t> cit(0);
            rocprim::radix_sort_pairs_desc(s->sort_temp, s->sort_temp_bytes, s->sub_len,
                                           s->sub_len_sorted, cit, s->sub_order,
                                           sub_total, 0, 8, stream);
        }
        et.mark(stream);
        hipLaunchKernelGGL(k_msm_bucket_reduce, dim3((sub_total + tb - 1) / tb), dim3(tb),
                           0, stream, s->vals_out, s->sub_start, s->sub_len, s->sub_order,
                           sub_total, d_bases, s->partials2);
        HIP_CHECK(hipGetLastError());
        {
            // order heads by sub count desc: the handful of long-merge heads
            // (top-window buckets) otherwise serialize whole waves
            rocprim::counting_iterator<uint32_t> cit(0);
            rocprim::radix_sort_pairs_desc(s->sort_temp, s->sort_temp_bytes, s->nsub,
                                           s->nsub_sorted, cit, s->head_order, hc, 0,
                                           16, stream);
        }
        hipLaunchKernelGGL(k_msm_seg_merge, dim3((hc + tb - 1) / tb), dim3(tb), 0,
                           stream, s->keys_out, s->heads, s->sub_off, s->nsub,
                           s->head_order, hc, s->partials2, s->buckets, c);
        HIP_CHECK(hipGetLastError());
    } else {
        et.mark(stream);
    }
    }  // end radix-sort path
    et.mark(stream);
    uint32_t cw = (1u << (c - 1)) / chunk_sz;
    // fused suffix-scan fold (k_msm_window_fold) measured SLOWER than the
    // chunk+combine pair at proof sizes (368 vs 402 proofs/s): Hillis-Steele
    // does a full-width EC add per lane per level and each level carries a
    // barrier + LDS Jacobian traffic; kept behind RNG_MSM_FOLD=1.
    static int fold_env = [] {
        const char* e = getenv("RNG_MSM_FOLD");
        return e ? atoi(e) : 0;
    }();
    bool use_fold = fold_env && cw <= 256;
    if (!use_fold) {  // separate chunk pass feeding the combine
        hipLaunchKernelGGL(k_msm_window_chunks,
                           dim3((uint32_t)((nchunks + tb - 1) / tb)), dim3(tb), 0,
                           stream, s->buckets, c, G, chunk_sz, s->partials);
        HIP_CHECK(hipGetLastError());
    }
    et.mark(stream);
    uint32_t subb = 1;
    if (use_fold) {
        if ((uint64_t)G > s->window_sums_cap) {
            hipFree(s->window_sums);
            s->window_sums_cap = (uint64_t)G * 2;
            HIP_CHECK(hipMalloc(&s->window_sums, s->window_sums_cap * sizeof(G1Jac)));
        }
        hipLaunchKernelGGL(k_msm_window_fold, dim3(G), dim3(cw), 0, stream,
                           s->buckets, c, chunk_sz, s->window_sums);
        HIP_CHECK(hipGetLastError());
        et.mark(stream);
    } else {
        // sub-blocks per window: enough blocks to spread chunks, but never
        // more than chunks (idle blocks); RNG_MSM_SUBB overrides for tuning
        subb = MSM_SUBB;
        while (subb > 1 && cw / subb < 64) subb >>= 1;
        // the combine blocks are serial EC chains: push toward >=2048 waves
        // (1 block = 1 wave of 64 lanes) while keeping >=64 chunks per block
        // so the scan-based combine2 applies
        while (subb < 256 && G * subb < 4096 && cw / (2 * subb) >= 64 &&
               (uint64_t)G * subb * 2 <= 8192)
            subb <<= 1;
        static int subb_env = [] {
            const char* e = getenv("RNG_MSM_SUBB");
            return e ? atoi(e) : 0;
        }();
        if (subb_env >= 1 && subb_env <= 256) subb = (uint32_t)subb_env;
        if ((uint64_t)G * subb > s->window_sums_cap) {
            // big fused batches (cohort k=64: G=10240 groups) outgrow the
            // default window_sums allocation
            hipFree(s->window_sums);
            s->window_sums_cap = (uint64_t)G * subb * 2;
            HIP_CHECK(hipMalloc(&s->window_sums, s->window_sums_cap * sizeof(G1Jac)));
        }
        static int msm_dbg = [] {
            const char* e = getenv("RNG_MSM_DEBUG");
            return e ? atoi(e) : 0;
        }();
        if (msm_dbg)
            fprintf(stderr, "[msm] n=%lu B=%u c=%u chunk=%u cw=%u subb=%u combine=%s\n",
                    (unsigned long)n, B, c, chunk_sz, cw, subb,
                    (cw % subb == 0 && (cw / subb) % 64 == 0) ? "scan2" : "v1");
        if (cw % subb == 0 && (cw / subb) % 64 == 0) {
            // scan-based combine: no per-chunk scalar weights
            hipLaunchKernelGGL(k_msm_window_combine2, dim3(G * subb), dim3(64), 0,
                               stream, s->partials, c, chunk_sz, subb,
                               s->window_sums);
        } else {
            hipLaunchKernelGGL(k_msm_window_combine, dim3(G * subb), dim3(64), 0,
                               stream, s->partials, c, chunk_sz, subb,
                               s->window_sums);
        }
        HIP_CHECK(hipGetLastError());
        et.mark(stream);
    }
    // host-side fold: G*SUBB Jacobians; a single-lane dependent EC chain is
    // far faster on a host core than on one GPU lane
    std::vector<G1Jac> wsums((size_t)G * subb);
    HIP_CHECK(hipMemcpyAsync(wsums.data(), s->window_sums, G * subb * sizeof(G1Jac),
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    et.collect(tls_msm_times, 5);
    // two pooled stages: per-(poly, window) sub-block reduction, then the
    // per-poly Horner across windows (the window count is small; a serial
    // 1024-add fold on one core showed up at B=1 with wide subb)
    std::vector<G1Jac> wred((size_t)G);
    auto red_one = [&](uint32_t g) {
        G1Jac sum = wsums[(size_t)g * subb];
        for (uint32_t i = 1; i < subb; ++i) sum = sum.add(wsums[(size_t)g * subb + i]);
        wred[g] = sum;
    };
    auto fold_one = [&](uint32_t b) {
        G1Jac acc = wred[(size_t)b * W + W - 1];
        for (int w = (int)W - 2; w >= 0; --w) {
            for (uint32_t k = 0; k < c; ++k) acc = acc.dbl();
            acc = acc.add(wred[(size_t)b * W + w]);
        }
        h_result[b] = acc;
    };
    if ((uint64_t)G * subb >= 64)
        HostPool::inst().parallel_for(G, red_one);
    else
        for (uint32_t g = 0; g < G; ++g) red_one(g);
    if (B >= 8)
        HostPool::inst().parallel_for(B, fold_one);
    else
        for (uint32_t b = 0; b < B; ++b) fold_one(b);
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

// ---------------- TurboPlonk prover (GPU) ----------------

static int ensure_coset_tables(RngCtxImpl* ctx, NttPlan* p) {
    std::lock_guard<std::mutex> lk(ctx->mu);  // thread-safe lazy build
    if (p->gpow) return RNG_OK;
    uint32_t m = p->n;
    Fr g = Fr::from_u64(FR_GENERATOR);
    Fr gi = g.inverse();
    if (build_pow_table(&p->gpow, g, 1, m) != RNG_OK) return RNG_ERR_HIP;
    if (build_pow_table(&p->gpow_inv, gi, 1, m) != RNG_OK) return RNG_ERR_HIP;
    Fr wm = h_fr_root_of_unity(m);
    if (build_pow_table(&p->xpow, wm, 1, m) != RNG_OK) return RNG_ERR_HIP;
    uint32_t blocks = (m + 255) / 256;
    hipLaunchKernelGGL(k_scale_const, dim3(blocks), dim3(256), 0, 0, p->xpow, g, m);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipDeviceSynchronize());
    return RNG_OK;
}

// forward coset NTT of device buffer `in` (m elements, coefficient form),
// result into `out` (in is clobbered)
static int coset_fwd_dev(RngCtxImpl* ctx, Fr* in, Fr* out, uint32_t m) {
    NttPlan* p = get_plan(ctx, m, 1);
    if (!p) return RNG_ERR_HIP;
    if (ensure_coset_tables(ctx, p) != RNG_OK) return RNG_ERR_HIP;
    uint32_t blocks = (m + 255) / 256;
    hipLaunchKernelGGL(k_mul_pointwise, dim3(blocks), dim3(256), 0, RNG_STREAM, in,
                       p->gpow, m);
    HIP_CHECK(hipGetLastError());
    return ntt_dev_run(ctx, in, out, m, 1, false, RNG_STREAM);
}

// inverse coset NTT of `in` (m evals), result into `out`
static int coset_inv_dev(RngCtxImpl* ctx, Fr* in, Fr* out, uint32_t m) {
    NttPlan* p = get_plan(ctx, m, 1);
    if (!p) return RNG_ERR_HIP;
    if (ensure_coset_tables(ctx, p) != RNG_OK) return RNG_ERR_HIP;
    int rc = ntt_dev_run(ctx, in, out, m, 1, true, RNG_STREAM);
    if (rc != RNG_OK) return rc;
    uint32_t blocks = (m + 255) / 256;
    hipLaunchKernelGGL(k_mul_pointwise, dim3(blocks), dim3(256), 0, RNG_STREAM, out,
                       p->gpow_inv, m);
    HIP_CHECK(hipGetLastError());
    return RNG_OK;
}

struct ProveScratch {  // per-context device scratch for proving at size n
    uint64_t n = 0, m = 0;
    Fr* w_coset = nullptr;   // 5*m
    Fr* z_coset = nullptr;   // m
    Fr* pi_coset = nullptr;  // m
    Fr* q_buf = nullptr;     // m
    Fr* tmp = nullptr;       // m
    Fr* stage = nullptr;     // n+3 coefficient staging
    uint64_t* canon = nullptr;  // 4*(n+3) canonical scalars for commits
    ~ProveScratch() {
        for (void* b : {(void*)w_coset, (void*)z_coset, (void*)pi_coset, (void*)q_buf,
                        (void*)tmp, (void*)stage, (void*)canon})
            hip_free_guarded(b);
    }
};

struct PlonkPkImpl {
    uint64_t n = 0, npub = 0;
    Fr k[5];
    std::vector<Fr> selq[13], sigp[5];       // host coeffs
    std::vector<Fr> sig_evals[5];            // host evals (z build)
    G1Aff sel_comms[13], sig_comms[5];
    bool sel_inf[13] = {false}, sig_inf[5] = {false};
    Fr* sel_coset = nullptr;   // device 13*m
    Fr* sig_coset = nullptr;   // device 5*m
    Fr* l1_coset = nullptr;    // device m
    RngCtxImpl* ctx = nullptr;
    ~PlonkPkImpl() {
        for (void* b : {(void*)sel_coset, (void*)sig_coset, (void*)l1_coset})
            hip_free_guarded(b);
    }
};

static thread_local std::unique_ptr<ProveScratch> tls_prove_scratch;

static int prove_scratch_ensure(uint64_t n) {
    uint64_t m = 8 * n;
    if (tls_prove_scratch && tls_prove_scratch->n >= n) return RNG_OK;
    tls_prove_scratch = std::make_unique<ProveScratch>();
    ProveScratch* s = tls_prove_scratch.get();
    HIP_CHECK(hipMalloc(&s->w_coset, 5 * m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->z_coset, m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->pi_coset, m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->q_buf, m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->tmp, m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->stage, 13 * (n + 3) * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->canon, 13 * 4 * (n + 3) * 8));
    s->n = n;
    s->m = m;
    return RNG_OK;
}

// commit to B host coefficient vectors via ONE fused GPU MSM over the SRS
// bases (shorter polys zero-padded; zero scalars cost nothing)
static int commit_dev_batch(RngCtxImpl* ctx, const std::vector<Fr>* const* polys,
                            uint32_t B, G1Aff* out, bool* out_inf) {
    ProveScratch* s = tls_prove_scratch.get();
    uint64_t m = 0;
    for (uint32_t b = 0; b < B; ++b)
        if (polys[b]->size() > m) m = polys[b]->size();
    if (m > ctx->srs_count || B > 13) return RNG_ERR_BAD_ARG;
    HIP_CHECK(hipMemsetAsync(s->stage, 0, B * m * sizeof(Fr), RNG_STREAM));
    for (uint32_t b = 0; b < B; ++b)
        HIP_CHECK(hipMemcpyAsync(s->stage + b * m, polys[b]->data(),
                                 polys[b]->size() * sizeof(Fr), hipMemcpyHostToDevice,
                                 RNG_STREAM));
    uint32_t blocks = (uint32_t)((B * m + 255) / 256);
    hipLaunchKernelGGL(k_fr_to_canonical, dim3(blocks), dim3(256), 0, RNG_STREAM,
                       s->stage, s->canon, (uint32_t)(B * m));
    HIP_CHECK(hipGetLastError());
    G1Jac res[13];
    // window size: auto by problem size; RNG_MSM_C overrides for tuning
    static int c_env = [] {
        const char* e = getenv("RNG_MSM_C");
        return e ? atoi(e) : 0;
    }();
    uint32_t c_used = (c_env >= 8 && c_env <= 16) ? (uint32_t)c_env : msm_auto_c(m);
    int rc = msm_dev_run((const G1Aff*)ctx->srs_dev, s->canon, m, c_used, res, B,
                         RNG_STREAM, (const G1Aff*)ctx->srs_glv_dev);
    if (rc != RNG_OK) return rc;
    for (uint32_t b = 0; b < B; ++b) {
        uint64_t rec[9];
        jac_to_affine_record(res[b], rec);
        memcpy(out[b].x.l, rec, 32);
        memcpy(out[b].y.l, rec + 4, 32);
        out_inf[b] = rec[8] != 0;
    }
    return RNG_OK;
}

static int commit_dev(RngCtxImpl* ctx, const std::vector<Fr>& coeffs, G1Aff* out,
                      bool* out_inf) {
    const std::vector<Fr>* p = &coeffs;
    return commit_dev_batch(ctx, &p, 1, out, out_inf);
}

// ---------------- cohort proving ----------------
// The reference proves many jobs concurrently from a rayon pool
// (native_proof_manager.rs:143-148,193-198).  Per-proof GPU runs at n=4096
// are aggregation-latency chains (r01 profile: window_combine 30% +
// seg_merge 22% of proof-workload GPU time), so the cohort prover advances
// k proofs in LOCKSTEP — R1 for all, R2 for all, … — and fuses each round's
// commitments into ONE msm_dev_run (key group g = poly*W + window already
// supports a batch dimension).  Transcripts and blinders stay per-proof
// (each transcript consumes only its own commitments), so cohort proofs are
// bit-identical to rng_prove with the same seed.  Host-side phases (grand
// product, evaluations, linearization) run on a persistent worker pool
// (HostPool, defined before msm_dev_run so its window fold can use it).

static void h_transcript_init(HostTranscript& tr, const PlonkPkImpl& pk, const Fr* pubs);

struct CohortScratch {  // per-thread device scratch for cohort proving
    uint64_t n = 0, k = 0;
    Fr* stage = nullptr;     // max(5k*(n+3), k*m) staging for commits/NTTs
    uint64_t* canon = nullptr;  // canonical scalars for the fused MSM
    Fr* coset_in = nullptr;  // 7k*(n+3) coefficient staging for the coset batch
    Fr* coset_out = nullptr; // 7k*m coset evaluations (5 wires + z + PI per proof)
    Fr* q_all = nullptr;     // k*m quotient evals
    Fr* ntt_tmp = nullptr;   // 7k*m NTT ping-pong
    QuotChal* chs = nullptr;  // k challenge packs (batched quotient)
    Fr* blinders = nullptr;   // 10k wire + 4k chunk blinders (device)
    ~CohortScratch() {
        for (void* b : {(void*)stage, (void*)canon, (void*)coset_in, (void*)coset_out,
                        (void*)q_all, (void*)ntt_tmp, (void*)chs, (void*)blinders})
            hip_free_guarded(b);
    }
};

static thread_local std::unique_ptr<CohortScratch> tls_cohort_scratch;

static int cohort_scratch_ensure(uint64_t n, uint64_t k) {
    if (tls_cohort_scratch && tls_cohort_scratch->n >= n && tls_cohort_scratch->k >= k)
        return RNG_OK;
    // grow to the max of old/new in BOTH dimensions: a mixed-circuit stream
    // (service batcher: n=4096 settlements between n=16384 validities)
    // would otherwise thrash reallocations between shapes
    if (tls_cohort_scratch) {
        if (tls_cohort_scratch->n > n) n = tls_cohort_scratch->n;
        if (tls_cohort_scratch->k > k) k = tls_cohort_scratch->k;
    }
    uint64_t m = 8 * n;
    tls_cohort_scratch = std::make_unique<CohortScratch>();
    CohortScratch* s = tls_cohort_scratch.get();
    uint64_t stage_elems = 5 * k * (n + 3);
    if (stage_elems < k * m) stage_elems = k * m;
    HIP_CHECK(hipMalloc(&s->stage, stage_elems * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->canon, stage_elems * 4 * 8));
    HIP_CHECK(hipMalloc(&s->coset_in, 7 * k * (n + 3) * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->coset_out, 7 * k * m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->q_all, k * m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->ntt_tmp, 7 * k * m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&s->chs, k * sizeof(QuotChal)));
    HIP_CHECK(hipMalloc(&s->blinders, 14 * k * sizeof(Fr)));
    s->n = n;
    s->k = k;
    return RNG_OK;
}

static int commit_staged(RngCtxImpl* ctx, uint64_t m, uint32_t B, G1Aff* out,
                         bool* out_inf);

// Fused commitment of B host polynomials (any B; shorter polys zero-padded
// to the longest) through ONE msm_dev_run.  Host window fold + affine
// conversion are parallelized across the pool for large B.
static int commit_cohort(RngCtxImpl* ctx, const std::vector<Fr>* const* polys,
                         uint32_t B, G1Aff* out, bool* out_inf) {
    static int trace = [] {
        const char* e = getenv("RNG_COHORT_TRACE");
        return e ? atoi(e) : 0;
    }();
    auto tr = [&](const char* tag) {
        if (trace) {
            fprintf(stderr, "[commit_cohort B=%u] %s\n", B, tag);
            fflush(stderr);
        }
    };
    auto tr_msm_stages = [&] {
        if (trace)
            fprintf(stderr,
                    "[commit_cohort B=%u] msm stages ms: digits %.2f sort %.2f "
                    "bucket %.2f chunks %.2f final %.2f\n",
                    B, tls_msm_times[0], tls_msm_times[1], tls_msm_times[2],
                    tls_msm_times[3], tls_msm_times[4]);
    };
    tr("enter");
    CohortScratch* s = tls_cohort_scratch.get();
    uint64_t m = 0;
    for (uint32_t b = 0; b < B; ++b)
        if (polys[b]->size() > m) m = polys[b]->size();
    if (m > ctx->srs_count) return RNG_ERR_BAD_ARG;
    // pack on the host (pool) and ship ONE H2D copy: B small async copies of
    // pageable memory each pay a staging round trip.  Plain local (NOT
    // thread_local): pool workers reference it from their own threads.
    std::vector<Fr> packed(B * m);
    HostPool::inst().parallel_for(B, [&](uint32_t b) {
        memcpy(packed.data() + (size_t)b * m, polys[b]->data(),
               polys[b]->size() * sizeof(Fr));
        memset(packed.data() + (size_t)b * m + polys[b]->size(), 0,
               (m - polys[b]->size()) * sizeof(Fr));
    });
    HIP_CHECK(hipMemcpyAsync(s->stage, packed.data(), B * m * sizeof(Fr),
                             hipMemcpyHostToDevice, RNG_STREAM));
    tr("msm");
    int rc = commit_staged(ctx, m, B, out, out_inf);
    tr("msm-done");
    tr_msm_stages();
    return rc;
}

// commit B polynomials that are ALREADY staged on device in s->stage
// (B consecutive m-coefficient blocks, Montgomery form)
static int commit_staged(RngCtxImpl* ctx, uint64_t m, uint32_t B, G1Aff* out,
                         bool* out_inf) {
    CohortScratch* s = tls_cohort_scratch.get();
    if (m > ctx->srs_count) return RNG_ERR_BAD_ARG;
    uint32_t blocks = (uint32_t)((B * m + 255) / 256);
    hipLaunchKernelGGL(k_fr_to_canonical, dim3(blocks), dim3(256), 0, RNG_STREAM,
                       s->stage, s->canon, (uint32_t)(B * m));
    HIP_CHECK(hipGetLastError());
    // bucket space per key group is 2^(c-1); it must stay well below the
    // per-poly point count n or empty buckets dominate the fused batch
    static int c_env = [] {
        const char* e = getenv("RNG_MSM_C");
        return e ? atoi(e) : 0;
    }();
    uint32_t c = (c_env >= 8 && c_env <= 16) ? (uint32_t)c_env : msm_auto_c(m);
    if ((uint64_t)B * ((256 + c - 1) / c) > 60000) return RNG_ERR_BAD_ARG;  // g<<16 cap
    std::vector<G1Jac> res(B);
    int rc = msm_dev_run((const G1Aff*)ctx->srs_dev, s->canon, m, c, res.data(), B,
                         RNG_STREAM, (const G1Aff*)ctx->srs_glv_dev);
    if (rc != RNG_OK) return rc;
    HostPool::inst().parallel_for(B, [&](uint32_t b) {
        uint64_t rec[9];
        jac_to_affine_record(res[b], rec);
        memcpy(out[b].x.l, rec, 32);
        memcpy(out[b].y.l, rec + 4, 32);
        out_inf[b] = rec[8] != 0;
    });
    return RNG_OK;
}

// k proofs in lockstep under ONE proving key (the headline shape: a stream
// of same-circuit jobs).  Bit-identical to k calls of plonk_prove_impl with
// the same seeds; 4 fused MSM runs per cohort instead of 5k.
static int plonk_prove_cohort_impl(RngCtxImpl* ctx, const PlonkPkImpl& pk, uint32_t k,
                                   const Fr* wires_all, const Fr* pubs_all,
                                   const uint64_t* seeds, uint64_t* out_proofs,
                                   uint64_t* out_hints) {
    const uint64_t n = pk.n;
    const uint32_t m = (uint32_t)(8 * n);
    const uint64_t hint_u64s = 4 * (n + 2) + 9;
    static int trace = [] {
        const char* e = getenv("RNG_COHORT_TRACE");
        return e ? atoi(e) : 0;
    }();
    auto t_enter = std::chrono::steady_clock::now();
#define COHORT_TRACE(tag)                                                     \
    do {                                                                      \
        if (trace) {                                                          \
            double ms_ = std::chrono::duration<double, std::milli>(           \
                             std::chrono::steady_clock::now() - t_enter)      \
                             .count();                                        \
            fprintf(stderr, "[cohort +%7.2fms] %s\n", ms_, tag);              \
            fflush(stderr);                                                   \
        }                                                                     \
    } while (0)
    COHORT_TRACE("enter");
    if (!cosets_ok(n)) return RNG_ERR_BAD_ARG;
    if (cohort_scratch_ensure(n, k) != RNG_OK) return RNG_ERR_HIP;
    CohortScratch* cs = tls_cohort_scratch.get();
    HostPool& pool = HostPool::inst();
    Fr w = h_fr_root_of_unity((uint32_t)n);

    std::vector<HostDrbg> drbg;
    drbg.reserve(k);
    std::vector<HostTranscript> tr(k);
    for (uint32_t p = 0; p < k; ++p) {
        drbg.emplace_back(seeds[p]);
        h_transcript_init(tr[p], pk, pubs_all + (size_t)p * pk.npub);
    }
    std::vector<std::vector<Fr>> wpoly(5 * (size_t)k), zpoly(k), quot_chunks(5 * (size_t)k);
    std::vector<G1Aff> comms(13 * (size_t)k);
    std::vector<uint8_t> cinf(13 * (size_t)k, 0);
    std::vector<Fr> beta(k), gamma(k), alpha(k), zeta(k), vch(k);

    // scratch for commit outputs (bool arrays for commit_cohort)
    std::vector<G1Aff> cbuf(5 * (size_t)k);
    std::unique_ptr<bool[]> ibuf(new bool[5 * (size_t)k]);
    std::vector<const std::vector<Fr>*> ps(5 * (size_t)k);

    COHORT_TRACE("r1");
    // --- R1: wire polys (one batched IFFT + one fused MSM) ---
    {
        HIP_CHECK(hipMemcpyAsync(cs->stage, wires_all, (size_t)5 * k * n * sizeof(Fr),
                                 hipMemcpyHostToDevice, RNG_STREAM));
        COHORT_TRACE("r1-ifft");
        int rc = ntt_dev_run(ctx, cs->stage, cs->ntt_tmp, (uint32_t)n, 5 * (uint64_t)k,
                             true, RNG_STREAM);
        if (rc != RNG_OK) return rc;
        COHORT_TRACE("r1-ifft-done");
        // blind + pad ON DEVICE (blinders drawn host-side in the single-
        // proof DRBG order), commit straight from the staged device buffer;
        // the blinded coefficients stream back D2H on the same stream and
        // are complete once the commit's internal sync returns
        std::vector<Fr> blind_h(10 * (size_t)k);
        for (uint32_t p = 0; p < k; ++p)
            for (int j = 0; j < 5; ++j) {
                blind_h[10 * (size_t)p + 2 * j] = drbg[p].next();
                blind_h[10 * (size_t)p + 2 * j + 1] = drbg[p].next();
            }
        HIP_CHECK(hipMemcpyAsync(cs->blinders, blind_h.data(),
                                 blind_h.size() * sizeof(Fr), hipMemcpyHostToDevice,
                                 RNG_STREAM));
        uint64_t total1 = 5 * (uint64_t)k * (n + 3);
        hipLaunchKernelGGL(k_blind_wires_batch,
                           dim3((uint32_t)((total1 + 255) / 256)), dim3(256), 0,
                           RNG_STREAM, cs->ntt_tmp, cs->blinders, cs->stage,
                           cs->coset_in, (uint32_t)n, total1);
        HIP_CHECK(hipGetLastError());
        uint64_t wlen = 5 * (uint64_t)k * (n + 2);
        std::vector<Fr> host_flat(wlen);
        HIP_CHECK(hipMemcpyAsync(host_flat.data(), cs->stage, wlen * sizeof(Fr),
                                 hipMemcpyDeviceToHost, RNG_STREAM));
        COHORT_TRACE("r1-sync-done");
        if (commit_staged(ctx, n + 2, 5 * k, cbuf.data(), ibuf.get()) != RNG_OK)
            return RNG_ERR_HIP;
        pool.parallel_for(k, [&](uint32_t p) {
            for (int j = 0; j < 5; ++j) {
                auto& wp = wpoly[5 * (size_t)p + j];
                const Fr* src = host_flat.data() + ((size_t)5 * p + j) * (n + 2);
                wp.assign(src, src + (n + 2));
            }
        });
        COHORT_TRACE("r1-blind-done");
        pool.parallel_for(k, [&](uint32_t p) {
            for (int j = 0; j < 5; ++j) {
                comms[13 * (size_t)p + j] = cbuf[5 * (size_t)p + j];
                cinf[13 * (size_t)p + j] = ibuf[5 * (size_t)p + j] ? 1 : 0;
                tr[p].append_g1(comms[13 * (size_t)p + j],
                                cinf[13 * (size_t)p + j] != 0);
            }
            if (out_hints) {
                uint64_t* h = out_hints + (size_t)p * hint_u64s;
                memcpy(h, wpoly[5 * (size_t)p].data(), (n + 2) * sizeof(Fr));
                uint64_t* c = h + 4 * (n + 2);
                memcpy(c, comms[13 * (size_t)p].x.l, 32);
                memcpy(c + 4, comms[13 * (size_t)p].y.l, 32);
                c[8] = cinf[13 * (size_t)p] ? 1 : 0;
            }
            beta[p] = tr[p].challenge();
            gamma[p] = tr[p].challenge();
        });
    }

    COHORT_TRACE("r2");
    // --- R2: grand products (host pool) + batched IFFT of z and PI ---
    {
        std::vector<Fr> evals_flat(2 * (size_t)k * n);  // k z-cols + k PI-cols
        pool.parallel_for(k, [&](uint32_t p) {
            const Fr* wires = wires_all + (size_t)5 * p * n;
            std::vector<Fr> znum(n), zden(n);
            Fr wi = Fr::one();
            for (uint64_t i = 0; i < n; ++i) {
                Fr num = Fr::one(), den = Fr::one();
                for (int j = 0; j < 5; ++j) {
                    Fr wv = wires[(size_t)j * n + i];
                    num = num.mul(wv.add(beta[p].mul(pk.k[j]).mul(wi)).add(gamma[p]));
                    den = den.mul(wv.add(beta[p].mul(pk.sig_evals[j][i])).add(gamma[p]));
                }
                znum[i] = num;
                zden[i] = den;
                wi = wi.mul(w);
            }
            std::vector<Fr> zden_inv = hbatch_inverse(zden);
            Fr* ze = evals_flat.data() + (size_t)p * n;
            ze[0] = Fr::one();
            for (uint64_t i = 1; i < n; ++i)
                ze[i] = ze[i - 1].mul(znum[i - 1]).mul(zden_inv[i - 1]);
            Fr* pie = evals_flat.data() + ((size_t)k + p) * n;
            for (uint64_t i = 0; i < n; ++i) pie[i] = Fr::zero();
            const Fr* pubs = pubs_all + (size_t)p * pk.npub;
            for (uint64_t i = 0; i < pk.npub; ++i) pie[i] = pubs[i];
        });
        HIP_CHECK(hipMemcpyAsync(cs->stage, evals_flat.data(),
                                 2 * (size_t)k * n * sizeof(Fr), hipMemcpyHostToDevice,
                                 RNG_STREAM));
        int rc = ntt_dev_run(ctx, cs->stage, cs->ntt_tmp, (uint32_t)n, 2 * (uint64_t)k,
                             true, RNG_STREAM);
        if (rc != RNG_OK) return rc;
        // device-side z blinding + PI staging straight into the R3 coset
        // slots (the wires landed there in R1); blinders in the single-proof
        // DRBG order (b2, b3, b4)
        std::vector<Fr> zblind_h(3 * (size_t)k);
        for (uint32_t p = 0; p < k; ++p) {
            zblind_h[3 * (size_t)p] = drbg[p].next();
            zblind_h[3 * (size_t)p + 1] = drbg[p].next();
            zblind_h[3 * (size_t)p + 2] = drbg[p].next();
        }
        HIP_CHECK(hipMemcpyAsync(cs->blinders, zblind_h.data(),
                                 zblind_h.size() * sizeof(Fr),
                                 hipMemcpyHostToDevice, RNG_STREAM));
        uint64_t totz = (uint64_t)k * (n + 3);
        hipLaunchKernelGGL(k_blind_z_batch, dim3((uint32_t)((totz + 255) / 256)),
                           dim3(256), 0, RNG_STREAM, cs->ntt_tmp, cs->blinders,
                           cs->stage, cs->coset_in, (uint32_t)n, totz);
        HIP_CHECK(hipGetLastError());
        hipLaunchKernelGGL(k_pi_to_coset, dim3((uint32_t)((totz + 255) / 256)),
                           dim3(256), 0, RNG_STREAM,
                           cs->ntt_tmp + (size_t)k * n, cs->coset_in, (uint32_t)n,
                           totz);
        HIP_CHECK(hipGetLastError());
        std::vector<Fr> z_host(totz);
        HIP_CHECK(hipMemcpyAsync(z_host.data(), cs->stage, totz * sizeof(Fr),
                                 hipMemcpyDeviceToHost, RNG_STREAM));
        if (commit_staged(ctx, n + 3, k, cbuf.data(), ibuf.get()) != RNG_OK)
            return RNG_ERR_HIP;
        // commit_staged synchronized: z_host is complete
        pool.parallel_for(k, [&](uint32_t p) {
            const Fr* src = z_host.data() + (size_t)p * (n + 3);
            zpoly[p].assign(src, src + (n + 3));
            comms[13 * (size_t)p + 5] = cbuf[p];
            cinf[13 * (size_t)p + 5] = ibuf[p] ? 1 : 0;
            tr[p].append_g1(cbuf[p], ibuf[p]);
            alpha[p] = tr[p].challenge();
        });
    }

    COHORT_TRACE("r3");
    // --- R3: coset transforms (batched) + quotient + fused MSM ---
    {
        NttPlan* mp = get_plan(ctx, m, 0);  // tables only; batch NTTs below
        if (!mp || ensure_coset_tables(ctx, mp) != RNG_OK) return RNG_ERR_HIP;
        const uint64_t stride = n + 3;
        uint64_t total_in = 7 * (uint64_t)k * stride;
        hipLaunchKernelGGL(k_mul_pointwise_mod,
                           dim3((uint32_t)((total_in + 255) / 256)), dim3(256), 0,
                           RNG_STREAM, cs->coset_in, mp->gpow, (uint32_t)stride,
                           total_in);
        HIP_CHECK(hipGetLastError());
        uint64_t total_m = 7 * (uint64_t)k * m;
        hipLaunchKernelGGL(k_copy_pad_batch, dim3((uint32_t)((total_m + 255) / 256)),
                           dim3(256), 0, RNG_STREAM, cs->coset_in, (uint32_t)stride,
                           cs->ntt_tmp, m, total_m);
        HIP_CHECK(hipGetLastError());
        int rc = ntt_dev_run(ctx, cs->ntt_tmp, cs->coset_out, m, 7 * (uint64_t)k, false,
                             RNG_STREAM);
        if (rc != RNG_OK) return rc;

        QuotChal ch;
        for (int j = 0; j < 5; ++j) ch.k[j] = pk.k[j];
        {
            Fr g = Fr::from_u64(FR_GENERATOR);
            Fr gn = g.pow_u64(n);
            Fr w8 = h_fr_root_of_unity(m).pow_u64(n);
            std::vector<Fr> zh(8);
            Fr cur = gn;
            for (int t = 0; t < 8; ++t) {
                zh[t] = cur.sub(Fr::one());
                cur = cur.mul(w8);
            }
            zh = hbatch_inverse(zh);
            for (int t = 0; t < 8; ++t) ch.zh_inv[t] = zh[t];
        }
        // ONE batched quotient launch for all k proofs (a single proof's m
        // grid is latency-starved; 483 small launches were 14% of cohort GPU
        // time in the r02 profile)
        {
            std::vector<QuotChal> chs_h(k, ch);
            for (uint32_t p = 0; p < k; ++p) {
                chs_h[p].beta = beta[p];
                chs_h[p].gamma = gamma[p];
                chs_h[p].alpha = alpha[p];
                chs_h[p].alpha2 = alpha[p].sqr();
            }
            HIP_CHECK(hipMemcpyAsync(cs->chs, chs_h.data(), k * sizeof(QuotChal),
                                     hipMemcpyHostToDevice, RNG_STREAM));
            uint64_t total_kq = (uint64_t)k * m;
            hipLaunchKernelGGL(k_quotient_batch,
                               dim3((uint32_t)((total_kq + 255) / 256)), dim3(256), 0,
                               RNG_STREAM, pk.sel_coset, pk.sig_coset, cs->coset_out,
                               pk.l1_coset, mp->xpow, cs->q_all, m, k, cs->chs);
            HIP_CHECK(hipGetLastError());
            HIP_CHECK(hipStreamSynchronize(RNG_STREAM));  // chs_h lifetime
        }
        // batched inverse coset NTT of the k quotients
        rc = ntt_dev_run(ctx, cs->q_all, cs->ntt_tmp, m, k, true, RNG_STREAM);
        if (rc != RNG_OK) return rc;
        uint64_t total_q = (uint64_t)k * m;
        hipLaunchKernelGGL(k_mul_pointwise_mod, dim3((uint32_t)((total_q + 255) / 256)),
                           dim3(256), 0, RNG_STREAM, cs->ntt_tmp, mp->gpow_inv, m,
                           total_q);
        HIP_CHECK(hipGetLastError());
        // chunk-split + linking blinders ON DEVICE, commit from the staged
        // buffer; the blinded chunks stream back D2H for the host R5
        // linearization (complete after the commit's internal sync)
        std::vector<Fr> qblind_h(4 * (size_t)k);
        for (uint32_t p = 0; p < k; ++p)
            for (int i = 0; i < 4; ++i) qblind_h[4 * (size_t)p + i] = drbg[p].next();
        HIP_CHECK(hipMemcpyAsync(cs->blinders, qblind_h.data(),
                                 qblind_h.size() * sizeof(Fr),
                                 hipMemcpyHostToDevice, RNG_STREAM));
        uint64_t total_c = 5 * (uint64_t)k * (n + 3);
        hipLaunchKernelGGL(k_quot_chunks_batch,
                           dim3((uint32_t)((total_c + 255) / 256)), dim3(256), 0,
                           RNG_STREAM, cs->ntt_tmp, cs->blinders, cs->stage,
                           (uint32_t)n, m, total_c);
        HIP_CHECK(hipGetLastError());
        std::vector<Fr> chunks_host(total_c);
        HIP_CHECK(hipMemcpyAsync(chunks_host.data(), cs->stage,
                                 total_c * sizeof(Fr), hipMemcpyDeviceToHost,
                                 RNG_STREAM));
        if (commit_staged(ctx, n + 3, 5 * k, cbuf.data(), ibuf.get()) != RNG_OK)
            return RNG_ERR_HIP;
        pool.parallel_for(k, [&](uint32_t p) {
            for (int i = 0; i < 5; ++i) {
                auto& qc = quot_chunks[5 * (size_t)p + i];
                const Fr* src = chunks_host.data() + (5 * (size_t)p + i) * (n + 3);
                qc.assign(src, src + (i < 4 ? n + 3 : n + 2));
            }
        });
        pool.parallel_for(k, [&](uint32_t p) {
            for (int i = 0; i < 5; ++i) {
                comms[13 * (size_t)p + 6 + i] = cbuf[5 * (size_t)p + i];
                cinf[13 * (size_t)p + 6 + i] = ibuf[5 * (size_t)p + i] ? 1 : 0;
                tr[p].append_g1(cbuf[5 * (size_t)p + i], ibuf[5 * (size_t)p + i]);
            }
            zeta[p] = tr[p].challenge();
        });
    }

    COHORT_TRACE("r4r5");
    // --- R4 + R5: evaluations, linearization, openings (host pool) ---
    std::vector<std::vector<Fr>> Wz(k), Wzw(k);
    std::vector<Fr> wire_evals(5 * (size_t)k), sigma_evals(4 * (size_t)k), zshift(k);
    pool.parallel_for(k, [&](uint32_t p) {
        Fr we[5], se[4];
        for (int j = 0; j < 5; ++j) {
            we[j] = hpoly_eval(wpoly[5 * (size_t)p + j], zeta[p]);
            tr[p].append_fr(we[j]);
            wire_evals[5 * (size_t)p + j] = we[j];
        }
        for (int j = 0; j < 4; ++j) {
            se[j] = hpoly_eval(pk.sigp[j], zeta[p]);
            tr[p].append_fr(se[j]);
            sigma_evals[4 * (size_t)p + j] = se[j];
        }
        Fr zs = hpoly_eval(zpoly[p], zeta[p].mul(w));
        zshift[p] = zs;
        tr[p].append_fr(zs);
        Fr v = tr[p].challenge();
        vch[p] = v;

        Fr zeta_n = zeta[p].pow_u64(n);
        Fr zh_zeta = zeta_n.sub(Fr::one());
        Fr l1_zeta = zh_zeta.mul(Fr::from_u64(n).mul(zeta[p].sub(Fr::one())).inverse());
        auto p5f = [](const Fr& x) {
            Fr x2 = x.sqr();
            return x2.sqr().mul(x);
        };
        const Fr* wb = we;
        std::vector<Fr> D;
        hpoly_add_scaled(D, pk.selq[11], Fr::one());
        for (int j = 0; j < 4; ++j) hpoly_add_scaled(D, pk.selq[j], wb[j]);
        hpoly_add_scaled(D, pk.selq[4], wb[0].mul(wb[1]));
        hpoly_add_scaled(D, pk.selq[5], wb[2].mul(wb[3]));
        for (int j = 0; j < 4; ++j) hpoly_add_scaled(D, pk.selq[6 + j], p5f(wb[j]));
        hpoly_add_scaled(D, pk.selq[12],
                         wb[0].mul(wb[1]).mul(wb[2]).mul(wb[3]).mul(wb[4]));
        hpoly_add_scaled(D, pk.selq[10], wb[4].neg());
        Fr fbar = Fr::one(), Bbar = Fr::one();
        for (int j = 0; j < 5; ++j)
            fbar = fbar.mul(wb[j].add(beta[p].mul(pk.k[j]).mul(zeta[p])).add(gamma[p]));
        for (int j = 0; j < 4; ++j)
            Bbar = Bbar.mul(wb[j].add(beta[p].mul(se[j])).add(gamma[p]));
        hpoly_add_scaled(D, zpoly[p],
                         alpha[p].mul(fbar).add(alpha[p].sqr().mul(l1_zeta)));
        hpoly_add_scaled(D, pk.sigp[4],
                         alpha[p].mul(beta[p]).mul(zs).mul(Bbar).neg());
        {
            Fr zpow = zh_zeta.neg();
            Fr step = zeta[p].pow_u64(n + 2);
            for (int i = 0; i < 5; ++i) {
                hpoly_add_scaled(D, quot_chunks[5 * (size_t)p + i], zpow);
                zpow = zpow.mul(step);
            }
        }
        std::vector<Fr> C = D;
        Fr vp = Fr::one();
        for (int j = 0; j < 5; ++j) {
            vp = vp.mul(v);
            hpoly_add_scaled(C, wpoly[5 * (size_t)p + j], vp);
        }
        for (int j = 0; j < 4; ++j) {
            vp = vp.mul(v);
            hpoly_add_scaled(C, pk.sigp[j], vp);
        }
        Wz[p] = hpoly_div_linear(C, zeta[p]);
        Wzw[p] = hpoly_div_linear(zpoly[p], zeta[p].mul(w));
    });
    for (uint32_t p = 0; p < k; ++p) {
        ps[2 * (size_t)p] = &Wz[p];
        ps[2 * (size_t)p + 1] = &Wzw[p];
    }
    if (commit_cohort(ctx, ps.data(), 2 * k, cbuf.data(), ibuf.get()) != RNG_OK)
        return RNG_ERR_HIP;
    pool.parallel_for(k, [&](uint32_t p) {
        comms[13 * (size_t)p + 11] = cbuf[2 * (size_t)p];
        cinf[13 * (size_t)p + 11] = ibuf[2 * (size_t)p] ? 1 : 0;
        comms[13 * (size_t)p + 12] = cbuf[2 * (size_t)p + 1];
        cinf[13 * (size_t)p + 12] = ibuf[2 * (size_t)p + 1] ? 1 : 0;
        tr[p].append_g1(cbuf[2 * (size_t)p], ibuf[2 * (size_t)p]);
        tr[p].append_g1(cbuf[2 * (size_t)p + 1], ibuf[2 * (size_t)p + 1]);
        // --- serialize ---
        uint64_t* out_proof = out_proofs + (size_t)p * 157;
        for (int i = 0; i < 13; ++i) {
            memcpy(out_proof + 9 * i, comms[13 * (size_t)p + i].x.l, 32);
            memcpy(out_proof + 9 * i + 4, comms[13 * (size_t)p + i].y.l, 32);
            out_proof[9 * i + 8] = cinf[13 * (size_t)p + i] ? 1 : 0;
        }
        uint64_t* e = out_proof + 117;
        for (int i = 0; i < 5; ++i)
            memcpy(e + 4 * i, wire_evals[5 * (size_t)p + i].l, 32);
        for (int i = 0; i < 4; ++i)
            memcpy(e + 20 + 4 * i, sigma_evals[4 * (size_t)p + i].l, 32);
        memcpy(e + 36, zshift[p].l, 32);
    });
    return RNG_OK;
}

// upload host coeffs, pad to m, coset-forward into dst (device, m elems)
static int coset_of_coeffs(RngCtxImpl* ctx, const std::vector<Fr>& coeffs, Fr* dst,
                           uint32_t m) {
    ProveScratch* s = tls_prove_scratch.get();
    HIP_CHECK(hipMemcpyAsync(s->stage, coeffs.data(), coeffs.size() * sizeof(Fr),
                             hipMemcpyHostToDevice, RNG_STREAM));
    uint32_t blocks = (m + 255) / 256;
    hipLaunchKernelGGL(k_copy_pad, dim3(blocks), dim3(256), 0, RNG_STREAM, s->stage,
                       (uint32_t)coeffs.size(), s->tmp, m);
    HIP_CHECK(hipGetLastError());
    return coset_fwd_dev(ctx, s->tmp, dst, m);
}

// GPU batched IFFT of column-major evals (host) -> host coeff vectors
static int ifft_columns(RngCtxImpl* ctx, const Fr* host_evals, uint64_t n,
                        uint64_t ncols, std::vector<Fr>* out_cols) {
    Fr* d = nullptr;
    HIP_CHECK(hipMalloc(&d, 2 * n * ncols * sizeof(Fr)));
    if (hipMemcpyAsync(d, host_evals, n * ncols * sizeof(Fr), hipMemcpyHostToDevice,
                       RNG_STREAM) != hipSuccess) {
        hipFree(d);
        return RNG_ERR_HIP;
    }
    Fr* out = (n <= NTT_SMALL_MAX) ? d : d + n * ncols;
    int rc = ntt_dev_run(ctx, d, out, (uint32_t)n, ncols, true, RNG_STREAM);
    if (rc == RNG_OK) {
        for (uint64_t c = 0; c < ncols; ++c) {
            out_cols[c].resize(n);
            if (hipMemcpyAsync(out_cols[c].data(), out + c * n, n * sizeof(Fr),
                               hipMemcpyDeviceToHost, RNG_STREAM) != hipSuccess)
                rc = RNG_ERR_HIP;
        }
        if (hipStreamSynchronize(RNG_STREAM) != hipSuccess) rc = RNG_ERR_HIP;
    }
    hipFree(d);
    return rc;
}

static int plonk_preprocess_impl(RngCtxImpl* ctx, const RngCircuitDesc* d,
                                 PlonkPkImpl* pk) {
    const uint64_t n = d->n, m = 8 * n;
    pk->n = n;
    pk->npub = d->num_public;
    coset_ks(pk->k);
    if (!cosets_ok(n)) return RNG_ERR_BAD_ARG;
    if (prove_scratch_ensure(n) != RNG_OK) return RNG_ERR_HIP;

    // selector + sigma polynomials (GPU IFFT)
    if (ifft_columns(ctx, (const Fr*)d->selectors, n, 13, pk->selq) != RNG_OK)
        return RNG_ERR_HIP;
    // sigma evals from permutation indices
    Fr w = h_fr_root_of_unity((uint32_t)n);
    std::vector<Fr> wpow(n);
    wpow[0] = Fr::one();
    for (uint64_t i = 1; i < n; ++i) wpow[i] = wpow[i - 1].mul(w);
    std::vector<Fr> sig_evals_flat(5 * n);
    for (int j = 0; j < 5; ++j) {
        for (uint64_t i = 0; i < n; ++i) {
            uint64_t slot = d->sigma[(size_t)j * n + i];
            sig_evals_flat[(size_t)j * n + i] = pk->k[slot / n].mul(wpow[slot % n]);
        }
        pk->sig_evals[j].assign(sig_evals_flat.begin() + (size_t)j * n,
                                sig_evals_flat.begin() + (size_t)(j + 1) * n);
    }
    if (ifft_columns(ctx, sig_evals_flat.data(), n, 5, pk->sigp) != RNG_OK)
        return RNG_ERR_HIP;

    // commitments (two fused batch MSMs)
    {
        const std::vector<Fr>* ps[13];
        for (int s = 0; s < 13; ++s) ps[s] = &pk->selq[s];
        if (commit_dev_batch(ctx, ps, 13, pk->sel_comms, pk->sel_inf) != RNG_OK)
            return RNG_ERR_HIP;
        for (int j = 0; j < 5; ++j) ps[j] = &pk->sigp[j];
        if (commit_dev_batch(ctx, ps, 5, pk->sig_comms, pk->sig_inf) != RNG_OK)
            return RNG_ERR_HIP;
    }

    // coset caches
    HIP_CHECK(hipMalloc(&pk->sel_coset, 13 * m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&pk->sig_coset, 5 * m * sizeof(Fr)));
    HIP_CHECK(hipMalloc(&pk->l1_coset, m * sizeof(Fr)));
    for (int s = 0; s < 13; ++s)
        if (coset_of_coeffs(ctx, pk->selq[s], pk->sel_coset + (size_t)s * m, (uint32_t)m) !=
            RNG_OK)
            return RNG_ERR_HIP;
    for (int j = 0; j < 5; ++j)
        if (coset_of_coeffs(ctx, pk->sigp[j], pk->sig_coset + (size_t)j * m, (uint32_t)m) !=
            RNG_OK)
            return RNG_ERR_HIP;
    {
        // L1 = IFFT(e0)
        std::vector<Fr> e0(n, Fr::zero());
        e0[0] = Fr::one();
        std::vector<Fr> l1coef[1];
        if (ifft_columns(ctx, e0.data(), n, 1, l1coef) != RNG_OK) return RNG_ERR_HIP;
        if (coset_of_coeffs(ctx, l1coef[0], pk->l1_coset, (uint32_t)m) != RNG_OK)
            return RNG_ERR_HIP;
    }
    HIP_CHECK(hipDeviceSynchronize());
    return RNG_OK;
}

// transcript init shared with the verifier side (spec: oracle/transcript.hpp)
static void h_transcript_init(HostTranscript& tr, const PlonkPkImpl& pk, const Fr* pubs) {
    tr.append_u64(pk.n);
    tr.append_u64(pk.npub);
    for (int s = 0; s < 13; ++s) tr.append_g1(pk.sel_comms[s], pk.sel_inf[s]);
    for (int j = 0; j < 5; ++j) tr.append_g1(pk.sig_comms[j], pk.sig_inf[j]);
    for (uint64_t i = 0; i < pk.npub; ++i) tr.append_fr(pubs[i]);
}

static int plonk_prove_impl(RngCtxImpl* ctx, const PlonkPkImpl& pk, const Fr* wires,
                            const Fr* pubs, uint64_t seed, uint64_t* out_proof,
                            uint64_t* out_link_hint) {
    const uint64_t n = pk.n;
    const uint32_t m = (uint32_t)(8 * n);
    if (prove_scratch_ensure(n) != RNG_OK) return RNG_ERR_HIP;
    ProveScratch* sc = tls_prove_scratch.get();
    HostDrbg drbg(seed);
    HostTranscript tr;
    h_transcript_init(tr, pk, pubs);
    Fr w = h_fr_root_of_unity((uint32_t)n);

    G1Aff comms[13];
    bool comm_inf[13] = {false};

    // --- R1: wire polys ---
    std::vector<Fr> wpoly[5];
    if (ifft_columns(ctx, wires, n, 5, wpoly) != RNG_OK) return RNG_ERR_HIP;
    for (int j = 0; j < 5; ++j) {
        Fr b0 = drbg.next(), b1 = drbg.next();
        wpoly[j].resize(n + 2, Fr::zero());
        wpoly[j][0] = wpoly[j][0].sub(b0);
        wpoly[j][1] = wpoly[j][1].sub(b1);
        wpoly[j][n] = wpoly[j][n].add(b0);
        wpoly[j][n + 1] = wpoly[j][n + 1].add(b1);
    }
    {
        const std::vector<Fr>* ps[5] = {&wpoly[0], &wpoly[1], &wpoly[2], &wpoly[3],
                                        &wpoly[4]};
        if (commit_dev_batch(ctx, ps, 5, comms, comm_inf) != RNG_OK) return RNG_ERR_HIP;
    }
    for (int j = 0; j < 5; ++j) tr.append_g1(comms[j], comm_inf[j]);
    if (out_link_hint) {
        // hint = wire-0 polynomial (n+2 coeffs, Montgomery) + its commitment
        memcpy(out_link_hint, wpoly[0].data(), (n + 2) * sizeof(Fr));
        uint64_t* c = out_link_hint + 4 * (n + 2);
        memcpy(c, comms[0].x.l, 32);
        memcpy(c + 4, comms[0].y.l, 32);
        c[8] = comm_inf[0] ? 1 : 0;
    }
    Fr beta = tr.challenge();
    Fr gamma = tr.challenge();

    // --- R2: grand product ---
    std::vector<Fr> znum(n), zden(n);
    {
        Fr wi = Fr::one();
        for (uint64_t i = 0; i < n; ++i) {
            Fr num = Fr::one(), den = Fr::one();
            for (int j = 0; j < 5; ++j) {
                Fr wv = wires[(size_t)j * n + i];
                num = num.mul(wv.add(beta.mul(pk.k[j]).mul(wi)).add(gamma));
                den = den.mul(wv.add(beta.mul(pk.sig_evals[j][i])).add(gamma));
            }
            znum[i] = num;
            zden[i] = den;
            wi = wi.mul(w);
        }
    }
    std::vector<Fr> zden_inv = hbatch_inverse(zden);
    std::vector<Fr> zevals(n);
    zevals[0] = Fr::one();
    for (uint64_t i = 1; i < n; ++i)
        zevals[i] = zevals[i - 1].mul(znum[i - 1]).mul(zden_inv[i - 1]);
    std::vector<Fr> zpoly[1];
    if (ifft_columns(ctx, zevals.data(), n, 1, zpoly) != RNG_OK) return RNG_ERR_HIP;
    {
        Fr b2 = drbg.next(), b3 = drbg.next(), b4 = drbg.next();
        zpoly[0].resize(n + 3, Fr::zero());
        zpoly[0][0] = zpoly[0][0].sub(b4);
        zpoly[0][1] = zpoly[0][1].sub(b3);
        zpoly[0][2] = zpoly[0][2].sub(b2);
        zpoly[0][n] = zpoly[0][n].add(b4);
        zpoly[0][n + 1] = zpoly[0][n + 1].add(b3);
        zpoly[0][n + 2] = zpoly[0][n + 2].add(b2);
    }
    if (commit_dev(ctx, zpoly[0], &comms[5], &comm_inf[5]) != RNG_OK) return RNG_ERR_HIP;
    tr.append_g1(comms[5], comm_inf[5]);
    Fr alpha = tr.challenge();

    // --- R3: quotient on the 8n coset (GPU) ---
    for (int j = 0; j < 5; ++j)
        if (coset_of_coeffs(ctx, wpoly[j], sc->w_coset + (size_t)j * m, m) != RNG_OK)
            return RNG_ERR_HIP;
    if (coset_of_coeffs(ctx, zpoly[0], sc->z_coset, m) != RNG_OK) return RNG_ERR_HIP;
    {
        std::vector<Fr> pie(n, Fr::zero());
        for (uint64_t i = 0; i < pk.npub; ++i) pie[i] = pubs[i];
        std::vector<Fr> picoef[1];
        if (ifft_columns(ctx, pie.data(), n, 1, picoef) != RNG_OK) return RNG_ERR_HIP;
        if (coset_of_coeffs(ctx, picoef[0], sc->pi_coset, m) != RNG_OK)
            return RNG_ERR_HIP;
    }
    QuotChal ch;
    ch.beta = beta;
    ch.gamma = gamma;
    ch.alpha = alpha;
    ch.alpha2 = alpha.sqr();
    for (int j = 0; j < 5; ++j) ch.k[j] = pk.k[j];
    {
        Fr g = Fr::from_u64(FR_GENERATOR);
        Fr gn = g.pow_u64(n);
        Fr w8 = h_fr_root_of_unity(m).pow_u64(n);
        std::vector<Fr> zh(8);
        Fr cur = gn;
        for (int t = 0; t < 8; ++t) {
            zh[t] = cur.sub(Fr::one());
            cur = cur.mul(w8);
        }
        zh = hbatch_inverse(zh);
        for (int t = 0; t < 8; ++t) ch.zh_inv[t] = zh[t];
    }
    NttPlan* mp = get_plan(ctx, m, 1);
    if (!mp || ensure_coset_tables(ctx, mp) != RNG_OK) return RNG_ERR_HIP;
    {
        uint32_t blocks = (m + 255) / 256;
        hipLaunchKernelGGL(k_quotient, dim3(blocks), dim3(256), 0, RNG_STREAM,
                           pk.sel_coset, pk.sig_coset, sc->w_coset, sc->z_coset,
                           sc->pi_coset, pk.l1_coset, mp->xpow, sc->q_buf, m, ch);
        HIP_CHECK(hipGetLastError());
    }
    if (coset_inv_dev(ctx, sc->q_buf, sc->tmp, m) != RNG_OK) return RNG_ERR_HIP;
    std::vector<Fr> quot(5 * (n + 2));
    HIP_CHECK(hipMemcpyAsync(quot.data(), sc->tmp, quot.size() * sizeof(Fr),
                             hipMemcpyDeviceToHost, RNG_STREAM));
    HIP_CHECK(hipStreamSynchronize(RNG_STREAM));
    std::vector<Fr> quot_chunks[5];
    {
        Fr prev = Fr::zero();
        for (int i = 0; i < 5; ++i) {
            quot_chunks[i].assign(quot.begin() + (size_t)i * (n + 2),
                                  quot.begin() + (size_t)(i + 1) * (n + 2));
            Fr bnext = (i < 4) ? drbg.next() : Fr::zero();
            quot_chunks[i][0] = quot_chunks[i][0].sub(prev);
            if (i < 4) {
                quot_chunks[i].resize(n + 3, Fr::zero());
                quot_chunks[i][n + 2] = quot_chunks[i][n + 2].add(bnext);
            }
            prev = bnext;
        }
        const std::vector<Fr>* ps[5] = {&quot_chunks[0], &quot_chunks[1], &quot_chunks[2],
                                        &quot_chunks[3], &quot_chunks[4]};
        if (commit_dev_batch(ctx, ps, 5, comms + 6, comm_inf + 6) != RNG_OK)
            return RNG_ERR_HIP;
        for (int i = 0; i < 5; ++i) tr.append_g1(comms[6 + i], comm_inf[6 + i]);
    }
    Fr zeta = tr.challenge();

    // --- R4: evaluations ---
    Fr wire_evals[5], sigma_evals[4], z_shift_eval;
    for (int j = 0; j < 5; ++j) {
        wire_evals[j] = hpoly_eval(wpoly[j], zeta);
        tr.append_fr(wire_evals[j]);
    }
    for (int j = 0; j < 4; ++j) {
        sigma_evals[j] = hpoly_eval(pk.sigp[j], zeta);
        tr.append_fr(sigma_evals[j]);
    }
    z_shift_eval = hpoly_eval(zpoly[0], zeta.mul(w));
    tr.append_fr(z_shift_eval);
    Fr v = tr.challenge();

    // --- R5: linearization + openings ---
    Fr zeta_n = zeta.pow_u64(n);
    Fr zh_zeta = zeta_n.sub(Fr::one());
    Fr l1_zeta = zh_zeta.mul(Fr::from_u64(n).mul(zeta.sub(Fr::one())).inverse());
    auto p5 = [](const Fr& x) {
        Fr x2 = x.sqr();
        return x2.sqr().mul(x);
    };
    const Fr* wb = wire_evals;
    std::vector<Fr> D;
    hpoly_add_scaled(D, pk.selq[11], Fr::one());
    for (int j = 0; j < 4; ++j) hpoly_add_scaled(D, pk.selq[j], wb[j]);
    hpoly_add_scaled(D, pk.selq[4], wb[0].mul(wb[1]));
    hpoly_add_scaled(D, pk.selq[5], wb[2].mul(wb[3]));
    for (int j = 0; j < 4; ++j) hpoly_add_scaled(D, pk.selq[6 + j], p5(wb[j]));
    hpoly_add_scaled(D, pk.selq[12],
                     wb[0].mul(wb[1]).mul(wb[2]).mul(wb[3]).mul(wb[4]));
    hpoly_add_scaled(D, pk.selq[10], wb[4].neg());
    Fr fbar = Fr::one(), Bbar = Fr::one();
    for (int j = 0; j < 5; ++j)
        fbar = fbar.mul(wb[j].add(beta.mul(pk.k[j]).mul(zeta)).add(gamma));
    for (int j = 0; j < 4; ++j)
        Bbar = Bbar.mul(wb[j].add(beta.mul(sigma_evals[j])).add(gamma));
    hpoly_add_scaled(D, zpoly[0], alpha.mul(fbar).add(alpha.sqr().mul(l1_zeta)));
    hpoly_add_scaled(D, pk.sigp[4], alpha.mul(beta).mul(z_shift_eval).mul(Bbar).neg());
    {
        Fr zpow = zh_zeta.neg();
        Fr step = zeta.pow_u64(n + 2);
        for (int i = 0; i < 5; ++i) {
            hpoly_add_scaled(D, quot_chunks[i], zpow);
            zpow = zpow.mul(step);
        }
    }
    std::vector<Fr> C = D;
    Fr vp = Fr::one();
    for (int j = 0; j < 5; ++j) {
        vp = vp.mul(v);
        hpoly_add_scaled(C, wpoly[j], vp);
    }
    for (int j = 0; j < 4; ++j) {
        vp = vp.mul(v);
        hpoly_add_scaled(C, pk.sigp[j], vp);
    }
    std::vector<Fr> Wz = hpoly_div_linear(C, zeta);
    std::vector<Fr> Wzw = hpoly_div_linear(zpoly[0], zeta.mul(w));
    {
        const std::vector<Fr>* ps[2] = {&Wz, &Wzw};
        if (commit_dev_batch(ctx, ps, 2, comms + 11, comm_inf + 11) != RNG_OK)
            return RNG_ERR_HIP;
    }
    tr.append_g1(comms[11], comm_inf[11]);
    tr.append_g1(comms[12], comm_inf[12]);

    // --- serialize (layout: include/rng_prover.h) ---
    for (int i = 0; i < 13; ++i) {
        memcpy(out_proof + 9 * i, comms[i].x.l, 32);
        memcpy(out_proof + 9 * i + 4, comms[i].y.l, 32);
        out_proof[9 * i + 8] = comm_inf[i] ? 1 : 0;
    }
    uint64_t* e = out_proof + 117;
    for (int i = 0; i < 5; ++i) memcpy(e + 4 * i, wire_evals[i].l, 32);
    for (int i = 0; i < 4; ++i) memcpy(e + 20 + 4 * i, sigma_evals[i].l, 32);
    memcpy(e + 36, z_shift_eval.l, 32);
    return RNG_OK;
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

/* Join the host worker pool's threads (harmless if never started).  Call
 * before process exit under a profiler: rocprofv3's finalizer can stall on
 * live foreign threads.  Subsequent pool work runs serially. */
void rng_shutdown_pool(void) { HostPool::inst().stop(); }

int rng_set_device(int device) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    hipError_t e = hipSetDevice(device);
    (void)hipGetLastError();  // clear the sticky last-error slot on failure
    return e == hipSuccess ? RNG_OK : RNG_ERR_HIP;
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
        // GLV: precompute the interleaved [P, phi(P)] array once (32 MB at 2^17)
        if (hipMalloc(&im->srs_glv_dev, npoints * 128) != hipSuccess) return nullptr;
        uint32_t blocks = (uint32_t)((npoints + 255) / 256);
        hipLaunchKernelGGL(k_bases_interleave, dim3(blocks), dim3(256), 0, 0,
                           (const G1Aff*)im->srs_dev, (G1Aff*)im->srs_glv_dev,
                           (uint32_t)npoints);
        if (hipDeviceSynchronize() != hipSuccess) return nullptr;
    }
    return ctx.release();
}

void rng_ctx_free(RngCtx* ctx) { delete ctx; }

// ---------------- deterministic TEST SRS (product-side) ----------------
// Dev/test deployments need a well-formed SRS without shipping ceremony
// bytes; generating it HERE means the daemon never loads oracle code
// (tier contract: oracle = test infrastructure only).  Byte-identical to
// the test oracle's generator (same tau derivation, same snarkjs subset
// layout parse_ptau_file reads — srs.rs:63-214); production passes real
// ptau bytes to rng_prover_init instead.

namespace {
// minimal host G2 Jacobian over Fq2 (EFD a=0 short-Weierstrass formulas,
// the same shapes as G1Jac) — only used for beta_h = tau*H at generation
struct HG2Jac {
    PFq2 X, Y, Z;
    static HG2Jac from_xy(const PFq2& x, const PFq2& y) { return {x, y, PFq2::one()}; }
    bool is_identity() const { return Z.eq(PFq2::zero()); }
    HG2Jac dbl() const {
        if (is_identity()) return *this;
        PFq2 A = X.sqr(), B = Y.sqr(), C = B.sqr();
        PFq2 D = X.add(B).sqr().sub(A).sub(C);
        D = D.add(D);
        PFq2 E = A.add(A).add(A);
        PFq2 F = E.sqr();
        HG2Jac r;
        r.X = F.sub(D.add(D));
        PFq2 C8 = C.add(C); C8 = C8.add(C8); C8 = C8.add(C8);
        r.Y = E.mul(D.sub(r.X)).sub(C8);
        r.Z = Y.mul(Z); r.Z = r.Z.add(r.Z);
        return r;
    }
    HG2Jac add(const HG2Jac& o) const {
        if (is_identity()) return o;
        if (o.is_identity()) return *this;
        PFq2 Z1Z1 = Z.sqr(), Z2Z2 = o.Z.sqr();
        PFq2 U1 = X.mul(Z2Z2), U2 = o.X.mul(Z1Z1);
        PFq2 S1 = Y.mul(o.Z).mul(Z2Z2), S2 = o.Y.mul(Z).mul(Z1Z1);
        if (U1.eq(U2)) {
            if (S1.eq(S2)) return dbl();
            return {PFq2::one(), PFq2::one(), PFq2::zero()};
        }
        PFq2 H = U2.sub(U1);
        PFq2 I = H.add(H).sqr();
        PFq2 J = H.mul(I);
        PFq2 rr = S2.sub(S1); rr = rr.add(rr);
        PFq2 V = U1.mul(I);
        HG2Jac r;
        r.X = rr.sqr().sub(J).sub(V.add(V));
        PFq2 SJ = S1.mul(J);
        r.Y = rr.mul(V.sub(r.X)).sub(SJ.add(SJ));
        r.Z = Z.add(o.Z).sqr().sub(Z1Z1).sub(Z2Z2).mul(H);
        return r;
    }
};

uint64_t srs_test_ptau_npoints(int power) { return (1ull << power) + 3; }
}  // namespace

uint64_t rng_srs_test_ptau_size(int power) {
    if (power < 2 || power > 20) return 0;
    return 12 + (12 + 44) + (12 + srs_test_ptau_npoints(power) * 64) + (12 + 256);
}

int rng_srs_gen_test_ptau(int power, uint64_t seed, uint8_t* out, size_t out_len) {
    uint64_t need = rng_srs_test_ptau_size(power);
    if (!out || !need || out_len < need) return RNG_ERR_BAD_ARG;
    // tau = keccak256("renegade-amd-srs-tau" || le64(seed)) reduced mod r
    uint8_t msg[28];
    memcpy(msg, "renegade-amd-srs-tau", 20);
    memcpy(msg + 20, &seed, 8);
    uint8_t h32[32];
    keccak256_h(msg, 28, h32);
    u64 limbs[4];
    memcpy(limbs, h32, 32);
    Fr tau = Fr::from_canonical(limbs);
    const uint64_t npoints = srs_test_ptau_npoints(power);
    std::vector<std::array<u64, 4>> scalars(npoints);
    Fr acc = Fr::one();
    for (uint64_t i = 0; i < npoints; ++i) {
        acc.to_canonical(scalars[i].data());
        acc = acc.mul(tau);
    }
    // fixed-base doubling table G_b = 2^b * G
    std::vector<G1Jac> table(256);
    {
        G1Aff g;
        static const u64 gx[4] = G1_GEN_X_MONT, gy[4] = G1_GEN_Y_MONT;
        memcpy(g.x.l, gx, 32);
        memcpy(g.y.l, gy, 32);
        table[0] = G1Jac::from_affine(g);
    }
    for (int j = 1; j < 256; ++j) table[j] = table[j - 1].dbl();
    std::vector<G1Aff> pts(npoints);
    unsigned nthreads = std::thread::hardware_concurrency();
    if (nthreads == 0) nthreads = 1;
    if (nthreads > 64) nthreads = 64;
    std::vector<std::thread> ths;
    std::atomic<uint64_t> next{0};
    for (unsigned t = 0; t < nthreads; ++t)
        ths.emplace_back([&] {
            for (;;) {
                uint64_t lo = next.fetch_add(512);
                if (lo >= npoints) break;
                uint64_t hi = lo + 512 < npoints ? lo + 512 : npoints;
                for (uint64_t i = lo; i < hi; ++i) {
                    G1Jac p = G1Jac::identity();
                    const u64* s = scalars[i].data();
                    for (int b = 0; b < 256; ++b)
                        if ((s[b >> 6] >> (b & 63)) & 1) p = p.add(table[b]);
                    Fq zi = p.Z.inverse();
                    Fq zi2 = zi.sqr();
                    pts[i].x = p.X.mul(zi2);
                    pts[i].y = p.Y.mul(zi2.mul(zi));
                }
            }
        });
    for (auto& th : ths) th.join();
    // beta_h = tau * H (canonical double-and-add, MSB first)
    PFq2 hx, hy;
    {
        static const u64 xc0[4] = G2_GEN_X_C0_MONT, xc1[4] = G2_GEN_X_C1_MONT;
        static const u64 yc0[4] = G2_GEN_Y_C0_MONT, yc1[4] = G2_GEN_Y_C1_MONT;
        memcpy(hx.a.l, xc0, 32);
        memcpy(hx.b.l, xc1, 32);
        memcpy(hy.a.l, yc0, 32);
        memcpy(hy.b.l, yc1, 32);
    }
    u64 tc[4];
    tau.to_canonical(tc);
    HG2Jac hjac = HG2Jac::from_xy(hx, hy);
    HG2Jac bacc{PFq2::one(), PFq2::one(), PFq2::zero()};
    for (int i = 255; i >= 0; --i) {
        bacc = bacc.dbl();
        if ((tc[i >> 6] >> (i & 63)) & 1) bacc = bacc.add(hjac);
    }
    PFq2 bx, by;
    {
        PFq2 zi = bacc.Z.inverse();
        PFq2 zi2 = zi.sqr();
        bx = bacc.X.mul(zi2);
        by = bacc.Y.mul(zi2.mul(zi));
    }
    // --- serialize (layout of srs_to_ptau / parse_ptau_file) ---
    uint8_t* w = out;
    auto put = [&](const void* p, size_t n) { memcpy(w, p, n); w += n; };
    auto put_u32 = [&](uint32_t v) { put(&v, 4); };
    auto put_u64v = [&](uint64_t v) { put(&v, 8); };
    put("ptau", 4);
    put_u32(1);
    put_u32(11);
    put_u32(1);
    put_u64v(44);
    put_u32(32);
    static const u64 qmod[4] = FQ_MODULUS;
    put(qmod, 32);
    put_u32((uint32_t)power);
    put_u32((uint32_t)power);
    put_u32(2);
    put_u64v(npoints * 64);
    for (uint64_t i = 0; i < npoints; ++i) {
        put(pts[i].x.l, 32);
        put(pts[i].y.l, 32);
    }
    put_u32(3);
    put_u64v(256);
    put(hx.a.l, 32);
    put(hx.b.l, 32);
    put(hy.a.l, 32);
    put(hy.b.l, 32);
    put(bx.a.l, 32);
    put(bx.b.l, 32);
    put(by.a.l, 32);
    put(by.b.l, 32);
    return RNG_OK;
}

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
    if (n <= NTT_SMALL_MAX)
        return ntt_dev_run(&ctx->impl, data, data, (uint32_t)n, batch, inverse);
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
    Fr* out = (n <= NTT_SMALL_MAX) ? d : p->scratch;
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
    uint32_t c = window_c > 0 ? (uint32_t)window_c : msm_auto_c(n);
    if (c < 8 || c > 16) return RNG_ERR_BAD_ARG;
    G1Jac res;
    int rc = msm_dev_run((const G1Aff*)dev_bases, (const uint64_t*)dev_scalars, n, c, &res, 1);
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

// `Valid Balance Create` circuit builder (zk_circuits/valid_balance_create.rs)
// with fixed-seed witness/statement; returns finalized tables.
void* rng_circ_build_vbc(uint64_t seed) {
    try {
        VbcWitness w;
        VbcStatement st;
        vbc_build_witness_statement(seed, w, st);
        PlonkCircuit cs;
        vbc_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_vbc: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_vbc: %s\n", e.what());
        return nullptr;
    }
}

// `Intent And Balance Private Settlement` circuit builder (the VALID MATCH
// MPC successor; zk_circuits/settlement/intent_and_balance_private_settlement.rs)
void* rng_circ_build_settlement(uint64_t seed) {
    try {
        SettlementWitness w;
        SettlementStatement st;
        settlement_build_witness_statement(seed, w, st);
        PlonkCircuit cs;
        settlement_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_settlement: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_settlement: %s\n", e.what());
        return nullptr;
    }
}

// --- validity <-> settlement proof bundle builders ---
// (zk_circuits/validity_proofs/intent_and_balance.rs; the validity circuit's
//  two link groups are placed at the SETTLEMENT circuit's layout, :316-341.)

// settlement circuit whose pre-update shares come from the two validity
// witnesses of the same seed (the consistent production bundle)
void* rng_circ_build_settlement_bundle(uint64_t seed) {
    try {
        ValidityBundle b;
        validity_bundle_build(seed, b);
        PlonkCircuit cs;
        settlement_apply_constraints(cs, b.sw, b.sst);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_settlement_bundle: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_settlement_bundle: %s\n", e.what());
        return nullptr;
    }
}

// INTENT AND BALANCE VALIDITY circuit for party 0 or 1 of the seed's bundle
void* rng_circ_build_validity(uint64_t seed, uint64_t party) {
    try {
        ValidityBundle b;
        validity_bundle_build(seed, b);
        // read the settlement circuit's link-group placement (inherited layout)
        uint64_t align = 0;
        int64_t off[2] = {0, 0};
        {
            PlonkCircuit scs;
            settlement_apply_constraints(scs, b.sw, b.sst);
            CircuitTables stt = scs.finalize();
            const char* names[2] = {"intent_and_balance_settlement_party0",
                                    "intent_and_balance_settlement_party1"};
            for (auto& g : stt.link_groups)
                for (int i = 0; i < 2; ++i)
                    if (g.id == names[i]) {
                        align = g.alignment;
                        off[i] = (int64_t)g.offset;
                    }
        }
        int p = (int)(party & 1);
        PlonkCircuit cs;
        validity_apply_constraints(cs, b.vw[p], b.vst[p], (int)align, off[0], off[1]);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_validity: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_validity: %s\n", e.what());
        return nullptr;
    }
}

// OUTPUT BALANCE VALIDITY circuit for party 0 or 1 of the seed's bundle
// (validity_proofs/output_balance.rs; links at the settlement's
//  output_balance_settlement_party{0,1} layout)
void* rng_circ_build_ob_validity(uint64_t seed, uint64_t party) {
    try {
        ValidityBundle b;
        validity_bundle_build(seed, b);
        uint64_t align = 0;
        int64_t off[2] = {0, 0};
        {
            PlonkCircuit scs;
            settlement_apply_constraints(scs, b.sw, b.sst);
            CircuitTables stt = scs.finalize();
            const char* names[2] = {"output_balance_settlement_party0",
                                    "output_balance_settlement_party1"};
            for (auto& g : stt.link_groups)
                for (int i = 0; i < 2; ++i)
                    if (g.id == names[i]) {
                        align = g.alignment;
                        off[i] = (int64_t)g.offset;
                    }
        }
        int p = (int)(party & 1);
        PlonkCircuit cs;
        ob_validity_apply_constraints(cs, b.ow[p], b.ost[p], (int)align, off[0], off[1]);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_ob_validity: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_ob_validity: %s\n", e.what());
        return nullptr;
    }
}

// INTENT AND BALANCE PUBLIC SETTLEMENT circuit for party 0 of the seed's
// bundle (settlement/intent_and_balance_public_settlement.rs); its two link
// groups are placed at the private settlement's party-0 layout.
void* rng_circ_build_public_settlement(uint64_t seed) {
    try {
        ValidityBundle b;
        validity_bundle_build(seed, b);
        uint64_t align = 0;
        int64_t pg_off = 0, og_off = 0;
        {
            PlonkCircuit scs;
            settlement_apply_constraints(scs, b.sw, b.sst);
            CircuitTables stt = scs.finalize();
            for (auto& g : stt.link_groups) {
                if (g.id == "intent_and_balance_settlement_party0") {
                    align = g.alignment;
                    pg_off = (int64_t)g.offset;
                }
                if (g.id == "output_balance_settlement_party0") og_off = (int64_t)g.offset;
            }
        }
        PubSettlementStatement st;
        pub_settlement_statement_from_bundle(b, st);
        PlonkCircuit cs;
        pub_settlement_apply_constraints(cs, b.sw.p[0], st, (int)align, pg_off, og_off);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_public_settlement: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_public_settlement: %s\n", e.what());
        return nullptr;
    }
}

// INTENT ONLY PUBLIC SETTLEMENT circuit (intent_only_public_settlement.rs)
void* rng_circ_build_io_settlement(uint64_t seed) {
    try {
        IoValidityWitness vw;
        IoValidityStatement vs;
        IoSettlementStatement ss;
        io_bundle_build(seed, vw, vs, ss);
        PlonkCircuit cs;
        io_settlement_apply_constraints(cs, vw.intent, ss);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_io_settlement: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_io_settlement: %s\n", e.what());
        return nullptr;
    }
}

// INTENT ONLY VALIDITY circuit (validity_proofs/intent_only.rs); its link
// group inherits the INTENT ONLY PUBLIC SETTLEMENT placement
void* rng_circ_build_io_validity(uint64_t seed) {
    try {
        IoValidityWitness vw;
        IoValidityStatement vs;
        IoSettlementStatement ss;
        io_bundle_build(seed, vw, vs, ss);
        uint64_t align = 0;
        int64_t off = 0;
        {
            PlonkCircuit scs;
            io_settlement_apply_constraints(scs, vw.intent, ss);
            CircuitTables stt = scs.finalize();
            for (auto& g : stt.link_groups)
                if (g.id == "intent_only_settlement") {
                    align = g.alignment;
                    off = (int64_t)g.offset;
                }
        }
        PlonkCircuit cs;
        io_validity_apply_constraints(cs, vw, vs, (int)align, off);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_io_validity: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_io_validity: %s\n", e.what());
        return nullptr;
    }
}

// INTENT ONLY BOUNDED SETTLEMENT circuit (intent_only_bounded_settlement.rs)
void* rng_circ_build_io_bounded_settlement(uint64_t seed) {
    try {
        IoValidityWitness vw;
        IoValidityStatement vs;
        IoSettlementStatement ss;
        io_bundle_build(seed, vw, vs, ss);
        uint64_t align = 0;
        int64_t off = 0;
        {
            PlonkCircuit scs;
            io_settlement_apply_constraints(scs, vw.intent, ss);
            CircuitTables stt = scs.finalize();
            for (auto& g : stt.link_groups)
                if (g.id == "intent_only_settlement") {
                    align = g.alignment;
                    off = (int64_t)g.offset;
                }
        }
        IoBoundedStatement st;
        io_bounded_statement_build(seed, vw, ss, st);
        PlonkCircuit cs;
        io_bounded_apply_constraints(cs, vw.intent, st, (int)align, off);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_io_bounded_settlement: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_io_bounded_settlement: %s\n", e.what());
        return nullptr;
    }
}

// INTENT AND BALANCE BOUNDED SETTLEMENT circuit
// (intent_and_balance_bounded_settlement.rs; party-0 groups at the private
//  settlement layout)
void* rng_circ_build_ib_bounded_settlement(uint64_t seed) {
    try {
        ValidityBundle b;
        validity_bundle_build(seed, b);
        uint64_t align = 0;
        int64_t pg_off = 0, og_off = 0;
        {
            PlonkCircuit scs;
            settlement_apply_constraints(scs, b.sw, b.sst);
            CircuitTables stt = scs.finalize();
            for (auto& g : stt.link_groups) {
                if (g.id == "intent_and_balance_settlement_party0") {
                    align = g.alignment;
                    pg_off = (int64_t)g.offset;
                }
                if (g.id == "output_balance_settlement_party0") og_off = (int64_t)g.offset;
            }
        }
        IbBoundedStatement st;
        ib_bounded_statement_from_bundle(b, st);
        PlonkCircuit cs;
        ib_bounded_apply_constraints(cs, b.sw.p[0], st, (int)align, pg_off, og_off);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_ib_bounded_settlement: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_ib_bounded_settlement: %s\n", e.what());
        return nullptr;
    }
}

// INTENT ONLY FIRST FILL VALIDITY (validity_proofs/intent_only_first_fill.rs);
// shares the seed's intent with rng_circ_build_io_settlement so the proofs link
void* rng_circ_build_ioff(uint64_t seed) {
    try {
        IoffWitness w;
        IoffStatement st;
        ioff_build(seed, w, st);
        uint64_t align = 0;
        int64_t off = 0;
        {
            IoValidityWitness vw;
            IoValidityStatement vs;
            IoSettlementStatement ss;
            io_bundle_build(seed, vw, vs, ss);
            PlonkCircuit scs;
            io_settlement_apply_constraints(scs, vw.intent, ss);
            CircuitTables stt = scs.finalize();
            for (auto& g : stt.link_groups)
                if (g.id == "intent_only_settlement") {
                    align = g.alignment;
                    off = (int64_t)g.offset;
                }
        }
        PlonkCircuit cs;
        ioff_apply_constraints(cs, w, st, (int)align, off);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_ioff: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_ioff: %s\n", e.what());
        return nullptr;
    }
}

// Baby Jubjub native test shims (tests/test_jubjub.py pins the C++ curve
// arithmetic + Schnorr/ElGamal against an independent pure-Python bignum
// implementation).  Scalars: 4 u64 plain LE; points: x,y Montgomery Fr.
int rng_jj_mul(const uint64_t* scalar4, const uint64_t* px4, const uint64_t* py4,
               uint64_t* out_xy8) {
    JjPoint p;
    memcpy(p.x.l, px4, 32);
    memcpy(p.y.l, py4, 32);
    if (!jj_on_curve(p) && !(p.x.is_zero())) return RNG_ERR_BAD_ARG;
    JjPoint r = jj_mul(scalar4, p);
    memcpy(out_xy8, r.x.l, 32);
    memcpy(out_xy8 + 4, r.y.l, 32);
    return RNG_OK;
}

void rng_jj_base(uint64_t* out_xy8) {
    JjPoint b = jj_base();
    memcpy(out_xy8, b.x.l, 32);
    memcpy(out_xy8 + 4, b.y.l, 32);
}

// sign/verify round trip: sk, nonce k as plain scalars; msg = n Fr
// (Montgomery); out = s (4 u64 plain), R (8 u64 Montgomery xy)
int rng_jj_sign(const uint64_t* sk4, const uint64_t* k4, const uint64_t* msg,
                uint64_t n, uint64_t* out_s4, uint64_t* out_r8) {
    JjScalar sk, k;
    memcpy(sk.v, sk4, 32);
    memcpy(k.v, k4, 32);
    JjSignature sig = jj_sign(sk, k, (const Fr*)msg, n);
    memcpy(out_s4, sig.s.v, 32);
    memcpy(out_r8, sig.R.x.l, 32);
    memcpy(out_r8 + 4, sig.R.y.l, 32);
    return RNG_OK;
}

int rng_jj_verify(const uint64_t* vk8, const uint64_t* s4, const uint64_t* r8,
                  const uint64_t* msg, uint64_t n) {
    JjPoint vk;
    memcpy(vk.x.l, vk8, 32);
    memcpy(vk.y.l, vk8 + 4, 32);
    JjSignature sig;
    memcpy(sig.s.v, s4, 32);
    memcpy(sig.R.x.l, r8, 32);
    memcpy(sig.R.y.l, r8 + 4, 32);
    return jj_verify(vk, sig, (const Fr*)msg, n) ? 1 : 0;
}

int rng_jj_elgamal(const uint64_t* pk8, const uint64_t* k4, const uint64_t* msg12,
                   uint64_t* out_eph8, uint64_t* out_c12) {
    JjPoint pk;
    memcpy(pk.x.l, pk8, 32);
    memcpy(pk.y.l, pk8 + 4, 32);
    JjScalar k;
    memcpy(k.v, k4, 32);
    JjCiphertext<3> ct = jj_elgamal_encrypt<3>(pk, k, (const Fr*)msg12);
    memcpy(out_eph8, ct.ephemeral_key.x.l, 32);
    memcpy(out_eph8 + 4, ct.ephemeral_key.y.l, 32);
    memcpy(out_c12, ct.ciphertext, 96);
    return RNG_OK;
}

// GLV decomposition test shim (host mirror of k_glv_decompose's math):
// canonical scalar -> (|k1|, sign1, |k2|, sign2); tests pin it against the
// Python derivation in scripts/gen_glv_params.py
int rng_glv_decompose(const uint64_t* canon4, uint64_t* out_k1, uint64_t* out_sign1,
                      uint64_t* out_k2, uint64_t* out_sign2) {
    // reuse the device function's logic via a 1-element host-side emulation:
    // the kernel math is plain integer C++, so call it through a tiny lambda
    // duplicating the steps (kept in sync with k_glv_decompose).
    const u64 G1C[4] = GLV_G1;
    const u64 G2C[4] = GLV_G2;
    const u64 A1[2] = GLV_A1;
    const u64 B1[2] = GLV_B1;
    const u64 A2[2] = GLV_A2;
    const u64 B2[2] = GLV_B2;
    u64 k[4];
    memcpy(k, canon4, 32);
    u64 c1[2], c2[2];
    glv_mul_shift(G1C, k, c1);
    glv_mul_shift(G2C, k, c2);
    u64 s1[4], s2[4], S[4], k1[4];
    glv_mul128(c1, A1, s1);
    glv_mul128(c2, A2, s2);
    glv_add4(s1, s2, S);
    if (glv_cmp4(k, S) >= 0) {
        glv_sub4(k, S, k1);
        *out_sign1 = 0;
    } else {
        glv_sub4(S, k, k1);
        *out_sign1 = 1;
    }
    u64 t1[4], t2[4], k2[4];
    glv_mul128(c1, B1, t1);
    glv_mul128(c2, B2, t2);
    if (glv_cmp4(t1, t2) >= 0) {
        glv_sub4(t1, t2, k2);
        *out_sign2 = 0;
    } else {
        glv_sub4(t2, t1, k2);
        *out_sign2 = 1;
    }
    memcpy(out_k1, k1, 32);
    memcpy(out_k2, k2, 32);
    return RNG_OK;
}

// embedded-curve gadget self-tests (Schnorr / ElGamal over Baby Jubjub):
// native-sign -> in-circuit verify; returns the finalized tables or null if
// the circuit is unsatisfied (tamper != 0 flips a signature/ciphertext bit
// and MUST yield null)
void* rng_testcirc_schnorr(uint64_t seed, int tamper) {
    try {
        Lcg rng(seed);
        auto sc = [&]() {
            JjScalar s{{rng.next() | (rng.next() << 52),
                        rng.next() | (rng.next() << 52),
                        rng.next() | (rng.next() << 52), rng.next() & 0x3FFFFFFFFFFull}};
            return s;  // < 2^250 < l
        };
        JjScalar sk = sc(), k = sc();
        Fr msg[3] = {rng.fr(), rng.fr(), rng.fr()};
        JjPoint vk = jj_pubkey(sk);
        JjSignature sig = jj_sign(sk, k, msg, 3);
        if (!jj_verify(vk, sig, msg, 3)) {
            fprintf(stderr, "rng_testcirc_schnorr: native verify failed\n");
            return nullptr;
        }
        if (tamper) sig.s.v[0] ^= 1;
        PlonkCircuit cs;
        JjPointVars vkv{cs.create_variable(vk.x), cs.create_variable(vk.y)};
        JjPointVars Rv{cs.create_variable(sig.R.x), cs.create_variable(sig.R.y)};
        Fr s_fr = Fr::from_canonical(sig.s.v);
        Var sv = cs.create_variable(s_fr);
        std::vector<Var> mv;
        for (int i = 0; i < 3; ++i) mv.push_back(cs.create_variable(msg[i]));
        schnorr_verify_gadget(cs, vkv, Rv, sv, mv);
        std::string why;
        if (!cs.check_satisfied(&why)) return nullptr;
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_testcirc_schnorr: %s\n", e.what());
        return nullptr;
    }
}

void* rng_testcirc_elgamal(uint64_t seed, int tamper) {
    try {
        Lcg rng(seed);
        JjScalar dk{{rng.next() | (rng.next() << 52), rng.next() | (rng.next() << 52),
                     rng.next() | (rng.next() << 52), rng.next() & 0x3FFFFFFFFFFull}};
        JjScalar k{{rng.next() | (rng.next() << 52), rng.next() | (rng.next() << 52),
                    rng.next() | (rng.next() << 52), rng.next() & 0x3FFFFFFFFFFull}};
        JjPoint pk = jj_pubkey(dk);
        Fr msg[3] = {rng.fr(), rng.fr(), rng.fr()};
        JjCiphertext<3> ct = jj_elgamal_encrypt<3>(pk, k, msg);
        if (tamper) ct.ciphertext[1] = ct.ciphertext[1].add(Fr::one());
        PlonkCircuit cs;
        JjPointVars pkv{cs.create_variable(pk.x), cs.create_variable(pk.y)};
        Var kv = cs.create_variable(Fr::from_canonical(k.v));
        std::vector<Var> mv;
        for (int i = 0; i < 3; ++i) mv.push_back(cs.create_variable(msg[i]));
        JjPointVars eph;
        std::vector<Var> cipher;
        elgamal_encrypt_gadget(cs, pkv, kv, mv, eph, cipher);
        // constrain against the native ciphertext (the statement shape)
        Var ex = cs.create_public_variable(ct.ephemeral_key.x);
        Var ey = cs.create_public_variable(ct.ephemeral_key.y);
        cs.enforce_equal(eph.x, ex);
        cs.enforce_equal(eph.y, ey);
        for (int i = 0; i < 3; ++i) {
            Var ci = cs.create_public_variable(ct.ciphertext[i]);
            cs.enforce_equal(cipher[i], ci);
        }
        std::string why;
        if (!cs.check_satisfied(&why)) return nullptr;
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_testcirc_elgamal: %s\n", e.what());
        return nullptr;
    }
}

// ---- fee circuits (zk_circuits/fees/) ----

// VALID NOTE REDEMPTION (fees/valid_note_redemption.rs)
void* rng_circ_build_note_redemption(uint64_t seed) {
    try {
        NoteRedemptionWitness w;
        NoteRedemptionStatement st;
        note_redemption_build(seed, w, st);
        PlonkCircuit cs;
        note_redemption_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_note_redemption: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_note_redemption: %s\n", e.what());
        return nullptr;
    }
}

// shared fee-payment builder; variant: 0 = public relayer, 1 = public
// protocol, 2 = private relayer (fees/valid_{public,private}_{relayer,
// protocol}_fee_payment.rs; private protocol needs in-circuit ElGamal and
// lands with the embedded-curve gadget)
static void* build_fee_payment(uint64_t seed, int variant, const char* name) {
    try {
        VdWitness w;
        FeePaymentStatement st;
        Note note;
        int field = (variant == 1) ? 6 : 5;
        fee_payment_build(seed, field, w, st, note);
        std::vector<Fr> ss = {st.merkle_root, st.old_balance_nullifier,
                              st.new_balance_commitment, st.recovery_id,
                              st.new_fee_balance_share};
        int note_mode = (variant == 2) ? 1 : 0;
        if (note_mode == 0) {
            auto nv = note.to_scalars();
            ss.insert(ss.end(), nv.begin(), nv.end());
        } else {
            ss.push_back(note.receiver);
            ss.push_back(native_note_commitment(note));
        }
        PlonkCircuit cs;
        fee_payment_apply_constraints(cs, w, note.blinder, field, note_mode,
                                      /*check_receiver=*/variant != 1, ss);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "%s: %s\n", name, why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "%s: %s\n", name, e.what());
        return nullptr;
    }
}

void* rng_circ_build_fee_public_relayer(uint64_t seed) {
    return build_fee_payment(seed, 0, "rng_circ_build_fee_public_relayer");
}
void* rng_circ_build_fee_public_protocol(uint64_t seed) {
    return build_fee_payment(seed, 1, "rng_circ_build_fee_public_protocol");
}
void* rng_circ_build_fee_private_relayer(uint64_t seed) {
    return build_fee_payment(seed, 2, "rng_circ_build_fee_private_relayer");
}

// VALID PRIVATE PROTOCOL FEE PAYMENT (in-circuit ElGamal note encryption)
void* rng_circ_build_fee_private_protocol(uint64_t seed) {
    try {
        VdWitness w;
        FeePaymentStatement st;
        Note note;
        fee_payment_build(seed, 6, w, st, note);
        Lcg rng(seed ^ 0xE161A3A1E161A3A1ull);
        // protocol receiver comes from the statement, not the balance
        u64 al[4] = {rng.next() | (rng.next() << 53), rng.next() | (rng.next() << 53),
                     rng.next() & 0xFFFFFFFF, 0};
        note.receiver = Fr::from_canonical(al);
        JjScalar dk = jj_random_scalar(rng);
        JjScalar k = jj_random_scalar(rng);
        JjPoint pk = jj_pubkey(dk);
        Fr plain[3] = {note.mint, note.amount, note.blinder};
        JjCiphertext<3> ct = jj_elgamal_encrypt<3>(pk, k, plain);
        std::vector<Fr> ss = {st.merkle_root, st.old_balance_nullifier,
                              st.new_balance_commitment, st.recovery_id,
                              st.new_fee_balance_share, note.receiver,
                              native_note_commitment(note), ct.ephemeral_key.x,
                              ct.ephemeral_key.y, ct.ciphertext[0], ct.ciphertext[1],
                              ct.ciphertext[2], pk.x, pk.y};
        PlonkCircuit cs;
        fee_private_protocol_apply_constraints(cs, w, note.blinder,
                                               Fr::from_canonical(k.v), ss);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_fee_private_protocol: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_fee_private_protocol: %s\n", e.what());
        return nullptr;
    }
}

// INTENT AND BALANCE FIRST FILL VALIDITY for bundle party 0/1
// (validity_proofs/intent_and_balance_first_fill.rs; links at the private
//  settlement's party layouts)
void* rng_circ_build_ff_validity(uint64_t seed, uint64_t party) {
    try {
        ValidityBundle b;
        validity_bundle_build(seed, b);
        uint64_t align = 0;
        int64_t off[2] = {0, 0};
        {
            PlonkCircuit scs;
            settlement_apply_constraints(scs, b.sw, b.sst);
            CircuitTables stt = scs.finalize();
            const char* names[2] = {"intent_and_balance_settlement_party0",
                                    "intent_and_balance_settlement_party1"};
            for (auto& g : stt.link_groups)
                for (int i = 0; i < 2; ++i)
                    if (g.id == names[i]) {
                        align = g.alignment;
                        off[i] = (int64_t)g.offset;
                    }
        }
        int p = (int)(party & 1);
        FfWitness w;
        FfStatement st;
        ff_build(b, p, seed, w, st);
        PlonkCircuit cs;
        ff_apply_constraints(cs, w, st, (int)align, off[0], off[1]);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_ff_validity: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_ff_validity: %s\n", e.what());
        return nullptr;
    }
}

// NEW OUTPUT BALANCE VALIDITY (validity_proofs/new_output_balance.rs; links
// at the settlement's output-balance layouts)
void* rng_circ_build_nob_validity(uint64_t seed) {
    try {
        NobWitness w;
        NobStatement st;
        nob_build(seed, w, st);
        uint64_t align = 0;
        int64_t off[2] = {0, 0};
        {
            ValidityBundle b;
            validity_bundle_build(seed, b);
            PlonkCircuit scs;
            settlement_apply_constraints(scs, b.sw, b.sst);
            CircuitTables stt = scs.finalize();
            const char* names[2] = {"output_balance_settlement_party0",
                                    "output_balance_settlement_party1"};
            for (auto& g : stt.link_groups)
                for (int i = 0; i < 2; ++i)
                    if (g.id == names[i]) {
                        align = g.alignment;
                        off[i] = (int64_t)g.offset;
                    }
        }
        PlonkCircuit cs;
        nob_apply_constraints(cs, w, st, (int)align, off[0], off[1]);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_nob_validity: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_nob_validity: %s\n", e.what());
        return nullptr;
    }
}

// ---- link-group layouts of the placing circuits (computed once) ----
// The settlement circuits PLACE the groups; validity circuits inherit them.
// Layout depends only on circuit structure, so compute from a fixed seed.
struct LinkLayouts {
    uint64_t ib_align = 0;
    int64_t pg0 = 0, og0 = 0, pg1 = 0, og1 = 0;  // private-settlement groups
    uint64_t io_align = 0;
    int64_t io_off = 0;  // intent-only settlement group
};
static const LinkLayouts& link_layouts() {
    static LinkLayouts L;
    static std::once_flag once;
    std::call_once(once, [] {
        ValidityBundle b;
        validity_bundle_build(42, b);
        PlonkCircuit scs;
        settlement_apply_constraints(scs, b.sw, b.sst);
        CircuitTables t = scs.finalize();
        for (auto& g : t.link_groups) {
            if (g.id == "intent_and_balance_settlement_party0") {
                L.ib_align = g.alignment;
                L.pg0 = (int64_t)g.offset;
            } else if (g.id == "intent_and_balance_settlement_party1") {
                L.pg1 = (int64_t)g.offset;
            } else if (g.id == "output_balance_settlement_party0") {
                L.og0 = (int64_t)g.offset;
            } else if (g.id == "output_balance_settlement_party1") {
                L.og1 = (int64_t)g.offset;
            }
        }
        IoValidityWitness vw;
        IoValidityStatement vs;
        IoSettlementStatement ss;
        io_bundle_build(42, vw, vs, ss);
        PlonkCircuit c2;
        io_settlement_apply_constraints(c2, vw.intent, ss);
        CircuitTables t2 = c2.finalize();
        for (auto& g : t2.link_groups)
            if (g.id == "intent_only_settlement") {
                L.io_align = g.alignment;
                L.io_off = (int64_t)g.offset;
            }
    });
    return L;
}

// ---- witness/statement parsers (scalar cursor over the field orders the
//      route bodies use; see rng_prover.h kind table) ----
namespace {
struct Rd {
    const Fr* p;
    Fr f() { return *p++; }
    uint64_t u() {
        u64 l[4];
        (*p).to_canonical(l);
        ++p;
        return l[0];
    }
    void frs(Fr* dst, int n) {
        for (int i = 0; i < n; ++i) dst[i] = f();
    }
    Csprng cs() {
        Csprng c;
        c.seed = f();
        c.index = u();
        return c;
    }
    Balance bal() {
        Balance b;
        Fr s[8];
        frs(s, 8);
        b.from_scalars(s);
        return b;
    }
    Intent in() {
        Intent i;
        i.in_token = f();
        i.out_token = f();
        i.owner = f();
        i.min_price_repr = f();
        i.amount_in = f();
        return i;
    }
    PostMatchShare pms() {
        PostMatchShare s;
        s.relayer_fee_balance = f();
        s.protocol_fee_balance = f();
        s.amount = f();
        return s;
    }
    Obligation ob() {
        Obligation o;
        o.input_token = f();
        o.output_token = f();
        o.amount_in = f();
        o.amount_out = f();
        return o;
    }
    StateBalance sbal() {
        StateBalance s;
        s.recovery = cs();
        s.share = cs();
        s.inner = bal();
        frs(s.public_share, 8);
        return s;
    }
    StateIntent sint() {
        StateIntent s;
        s.recovery = cs();
        s.share = cs();
        s.inner = in();
        frs(s.public_share, 5);
        return s;
    }
    void opening(Fr* elems, bool* idx) {
        frs(elems, MERKLE_HEIGHT);
        for (int i = 0; i < MERKLE_HEIGHT; ++i) idx[i] = u() != 0;
    }
    JjSignature sig() {
        JjSignature s;
        Fr sf = f();
        sf.to_canonical(s.s.v);
        s.R.x = f();
        s.R.y = f();
        return s;
    }
    Note note() {
        Note n;
        n.mint = f();
        n.amount = f();
        n.receiver = f();
        n.blinder = f();
        return n;
    }
    SettlementParty party_no_ob() {  // public/bounded settlement witness (28)
        SettlementParty p;
        p.intent = in();
        p.pre_amount_share = f();
        p.input_balance = bal();
        p.pre_in_shares = pms();
        p.output_balance = bal();
        p.pre_out_shares = pms();
        return p;
    }
};
}  // namespace

namespace {
struct Wt {  // writer mirroring Rd
    Fr* p;
    void f(const Fr& v) { *p++ = v; }
    void u(uint64_t x) { *p++ = Fr::from_u64(x); }
    void frs(const Fr* src, int n) {
        for (int i = 0; i < n; ++i) f(src[i]);
    }
    void cs(const Csprng& c) {
        f(c.seed);
        u(c.index);
    }
    void bal(const Balance& b) {
        for (auto& s : b.to_scalars()) f(s);
    }
    void in(const Intent& i) {
        for (auto& s : i.to_scalars()) f(s);
    }
    void pms(const PostMatchShare& s_) {
        for (auto& s : s_.to_scalars()) f(s);
    }
    void sbal(const StateBalance& s_) {
        for (auto& s : s_.to_scalars()) f(s);
    }
    void sint(const StateIntent& s_) {
        for (auto& s : s_.to_scalars()) f(s);
    }
    void opening(const Fr* elems, const bool* idx) {
        frs(elems, MERKLE_HEIGHT);
        for (int i = 0; i < MERKLE_HEIGHT; ++i) u(idx[i] ? 1 : 0);
    }
    void sig(const JjSignature& s_) {
        f(Fr::from_canonical(s_.s.v));
        f(s_.R.x);
        f(s_.R.y);
    }
    void st(const std::vector<Fr>& v) {
        for (auto& s : v) f(s);
    }
};
}  // namespace

// per-kind witness/statement scalar counts (the route body sizes)
int rng_ws_sizes(int kind, uint64_t* out_nw, uint64_t* out_ns) {
    struct {
        int k, nw, ns;
    } T[] = {{1, 40, 8},  {2, 40, 8},  {3, 34, 3},  {4, 91, 10}, {5, 69, 11},
             {6, 39, 7},  {7, 14, 8},  {8, 74, 10}, {9, 51, 5},  {11, 28, 14},
             {12, 28, 16}, {13, 5, 6},  {14, 5, 9},  {15, 20, 6}, {16, 40, 9},
             {17, 40, 9}, {18, 41, 7}, {19, 42, 14}};
    for (auto& t : T)
        if (t.k == kind) {
            *out_nw = t.nw;
            *out_ns = t.ns;
            return 0;
        }
    return RNG_ERR_BAD_ARG;
}

// fixed-seed witness/statement generator for any kind (test vectors for the
// prover-service routes; same serialization rng_circ_from_scalars parses)
int rng_witness_statement(int kind, uint64_t seed, uint64_t* out_w, uint64_t* out_s) {
    try {
        Wt w{(Fr*)out_w};
        Wt s{(Fr*)out_s};
        switch (kind) {
            case 1: {
                VdWitness vw;
                VdStatement st;
                vd_build_witness_statement(seed, vw, st);
                w.sbal(vw.old_balance);
                w.opening(vw.opening_elems, vw.opening_indices);
                s.st(st.to_scalars());
                break;
            }
            case 2: {
                VdWitness vw;
                VwStatement st;
                vw_build_witness_statement(seed, vw, st);
                w.sbal(vw.old_balance);
                w.opening(vw.opening_elems, vw.opening_indices);
                s.st(st.to_scalars());
                break;
            }
            case 3: {
                VocWitness vw;
                VocStatement st;
                voc_build_witness_statement(seed, vw, st);
                w.sint(vw.old_intent);
                w.opening(vw.opening_elems, vw.opening_idx);
                s.st(st.to_scalars());
                break;
            }
            case 4: {
                ValidityBundle b;
                validity_bundle_build(seed, b);
                const ValidityWitness& vw = b.vw[0];
                w.sint(vw.old_intent);
                w.opening(vw.intent_opening_elems, vw.intent_opening_idx);
                w.in(vw.intent);
                w.f(vw.new_amount_public_share);
                w.sbal(vw.old_balance);
                w.opening(vw.balance_opening_elems, vw.balance_opening_idx);
                w.bal(vw.balance);
                w.pms(vw.post_match_balance_shares);
                s.st(b.vst[0].to_scalars());
                break;
            }
            case 5: {
                ValidityBundle b;
                validity_bundle_build(seed, b);
                FfWitness vw;
                FfStatement st;
                ff_build(b, 0, seed, vw, st);
                w.in(vw.intent);
                w.cs(vw.share_stream);
                w.cs(vw.recovery_stream);
                w.frs(vw.private_intent_shares, 5);
                w.f(vw.new_amount_public_share);
                w.sig(vw.sig);
                w.sbal(vw.old_balance);
                w.bal(vw.balance);
                w.pms(vw.post_match_balance_shares);
                w.opening(vw.opening_elems, vw.opening_idx);
                s.st(st.to_scalars());
                break;
            }
            case 6: {
                IoValidityWitness vw;
                IoValidityStatement vs;
                IoSettlementStatement ss;
                io_bundle_build(seed, vw, vs, ss);
                w.sint(vw.old_intent);
                w.opening(vw.opening_elems, vw.opening_idx);
                w.in(vw.intent);
                s.st(vs.to_scalars());
                break;
            }
            case 7: {
                IoffWitness vw;
                IoffStatement st;
                ioff_build(seed, vw, st);
                w.in(vw.intent);
                w.cs(vw.share_stream);
                w.cs(vw.recovery_stream);
                w.frs(vw.private_shares, 5);
                s.st(st.to_scalars());
                break;
            }
            case 8: {
                NobWitness vw;
                NobStatement st;
                nob_build(seed, vw, st);
                w.sbal(vw.new_balance);
                w.bal(vw.balance);
                w.pms(vw.post_match_balance_shares);
                w.sbal(vw.existing_balance);
                w.opening(vw.opening_elems, vw.opening_idx);
                w.sig(vw.sig);
                s.st(st.to_scalars());
                break;
            }
            case 9: {
                ValidityBundle b;
                validity_bundle_build(seed, b);
                const ObValidityWitness& vw = b.ow[0];
                w.sbal(vw.old_balance);
                w.opening(vw.opening_elems, vw.opening_idx);
                w.bal(vw.balance);
                w.pms(vw.post_match_balance_shares);
                s.st(b.ost[0].to_scalars());
                break;
            }
            case 11: {
                ValidityBundle b;
                validity_bundle_build(seed, b);
                const SettlementParty& p = b.sw.p[0];
                w.in(p.intent);
                w.f(p.pre_amount_share);
                w.bal(p.input_balance);
                w.pms(p.pre_in_shares);
                w.bal(p.output_balance);
                w.pms(p.pre_out_shares);
                PubSettlementStatement st;
                pub_settlement_statement_from_bundle(b, st);
                s.st(st.to_scalars());
                break;
            }
            case 12: {
                ValidityBundle b;
                validity_bundle_build(seed, b);
                const SettlementParty& p = b.sw.p[0];
                w.in(p.intent);
                w.f(p.pre_amount_share);
                w.bal(p.input_balance);
                w.pms(p.pre_in_shares);
                w.bal(p.output_balance);
                w.pms(p.pre_out_shares);
                IbBoundedStatement st;
                ib_bounded_statement_from_bundle(b, st);
                s.st(st.to_scalars());
                break;
            }
            case 13: {
                IoValidityWitness vw;
                IoValidityStatement vs;
                IoSettlementStatement ss;
                io_bundle_build(seed, vw, vs, ss);
                w.in(vw.intent);
                s.st(ss.to_scalars());
                break;
            }
            case 14: {
                IoValidityWitness vw;
                IoValidityStatement vs;
                IoSettlementStatement ss;
                io_bundle_build(seed, vw, vs, ss);
                IoBoundedStatement st;
                io_bounded_statement_build(seed, vw, ss, st);
                w.in(vw.intent);
                s.st(st.to_scalars());
                break;
            }
            case 15: {
                NoteRedemptionWitness vw;
                NoteRedemptionStatement st;
                note_redemption_build(seed, vw, st);
                w.opening(vw.opening_elems, vw.opening_idx);
                s.st(st.to_scalars());
                break;
            }
            case 16:
            case 17:
            case 18: {
                VdWitness vw;
                FeePaymentStatement st;
                Note note;
                int field = (kind == 17) ? 6 : 5;
                fee_payment_build(seed, field, vw, st, note);
                w.sbal(vw.old_balance);
                w.opening(vw.opening_elems, vw.opening_indices);
                std::vector<Fr> ss = {st.merkle_root, st.old_balance_nullifier,
                                      st.new_balance_commitment, st.recovery_id,
                                      st.new_fee_balance_share};
                if (kind == 18) {
                    w.f(note.blinder);
                    ss.push_back(note.receiver);
                    ss.push_back(native_note_commitment(note));
                } else {
                    auto nv = note.to_scalars();
                    ss.insert(ss.end(), nv.begin(), nv.end());
                }
                s.st(ss);
                break;
            }
            case 19: {
                VdWitness vw;
                FeePaymentStatement st;
                Note note;
                fee_payment_build(seed, 6, vw, st, note);
                Lcg rng(seed ^ 0xE161A3A1E161A3A1ull);
                u64 al[4] = {rng.next() | (rng.next() << 53),
                             rng.next() | (rng.next() << 53), rng.next() & 0xFFFFFFFF, 0};
                note.receiver = Fr::from_canonical(al);
                JjScalar dk = jj_random_scalar(rng);
                JjScalar k = jj_random_scalar(rng);
                JjPoint pk = jj_pubkey(dk);
                Fr plain[3] = {note.mint, note.amount, note.blinder};
                JjCiphertext<3> ct = jj_elgamal_encrypt<3>(pk, k, plain);
                w.sbal(vw.old_balance);
                w.opening(vw.opening_elems, vw.opening_indices);
                w.f(note.blinder);
                w.f(Fr::from_canonical(k.v));
                s.st({st.merkle_root, st.old_balance_nullifier, st.new_balance_commitment,
                      st.recovery_id, st.new_fee_balance_share, note.receiver,
                      native_note_commitment(note), ct.ephemeral_key.x,
                      ct.ephemeral_key.y, ct.ciphertext[0], ct.ciphertext[1],
                      ct.ciphertext[2], pk.x, pk.y});
                break;
            }
            default:
                return RNG_ERR_BAD_ARG;
        }
        return RNG_OK;
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_witness_statement: %s\n", e.what());
        return RNG_ERR_BAD_ARG;
    }
}

// Party-aware test vectors for the two per-party validity kinds (4 = intent-
// and-balance validity, 9 = output-balance validity): the private-settlement
// route takes FOUR external hints (party 0/1 of each), all from one seed's
// consistent bundle (native_proof_manager.rs:554-590).
int rng_witness_statement_party(int kind, uint64_t seed, uint64_t party,
                                uint64_t* out_w, uint64_t* out_s) {
    try {
        Wt w{(Fr*)out_w};
        Wt s{(Fr*)out_s};
        int p = (int)(party & 1);
        ValidityBundle b;
        validity_bundle_build(seed, b);
        if (kind == 4) {
            const ValidityWitness& vw = b.vw[p];
            w.sint(vw.old_intent);
            w.opening(vw.intent_opening_elems, vw.intent_opening_idx);
            w.in(vw.intent);
            w.f(vw.new_amount_public_share);
            w.sbal(vw.old_balance);
            w.opening(vw.balance_opening_elems, vw.balance_opening_idx);
            w.bal(vw.balance);
            w.pms(vw.post_match_balance_shares);
            s.st(b.vst[p].to_scalars());
        } else if (kind == 9) {
            const ObValidityWitness& vw = b.ow[p];
            w.sbal(vw.old_balance);
            w.opening(vw.opening_elems, vw.opening_idx);
            w.bal(vw.balance);
            w.pms(vw.post_match_balance_shares);
            s.st(b.ost[p].to_scalars());
        } else if (kind == 10) {
            // the BUNDLE's settlement witness/statement (64/17 scalars, same
            // order as rng_settlement_witness_statement).  Needed because
            // validity_bundle_build mutates the settlement witness after
            // building it (real Schnorr authorities on the input balances),
            // so only the bundle's version links against the bundle's
            // validity proofs.  `party` is ignored.
            const SettlementWitness& sw = b.sw;
            std::vector<Fr> ws;
            for (int i = 0; i < 2; ++i) {
                auto a = sw.p[i].obligation.to_scalars();
                auto bi = sw.p[i].intent.to_scalars();
                auto c = sw.p[i].input_balance.to_scalars();
                auto d = sw.p[i].pre_in_shares.to_scalars();
                auto e = sw.p[i].output_balance.to_scalars();
                auto f = sw.p[i].pre_out_shares.to_scalars();
                ws.insert(ws.end(), a.begin(), a.end());
                ws.insert(ws.end(), bi.begin(), bi.end());
                ws.push_back(sw.p[i].pre_amount_share);
                ws.insert(ws.end(), c.begin(), c.end());
                ws.insert(ws.end(), d.begin(), d.end());
                ws.insert(ws.end(), e.begin(), e.end());
                ws.insert(ws.end(), f.begin(), f.end());
            }
            auto ss = b.sst.to_scalars();
            memcpy(out_w, ws.data(), ws.size() * sizeof(Fr));
            memcpy(out_s, ss.data(), ss.size() * sizeof(Fr));
        } else {
            return RNG_ERR_BAD_ARG;
        }
        return RNG_OK;
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_witness_statement_party: %s\n", e.what());
        return RNG_ERR_BAD_ARG;
    }
}

// Build any circuit from caller-supplied witness/statement scalars (the
// prover-service request shape).  `kind` table in include/rng_prover.h;
// returns null if the witness does not satisfy the circuit.
void* rng_circ_from_scalars(int kind, const uint64_t* witness64,
                            const uint64_t* statement64) {
    try {
        Rd w{(const Fr*)witness64};
        Rd s{(const Fr*)statement64};
        const LinkLayouts& L = link_layouts();
        PlonkCircuit cs;
        switch (kind) {
            case 1:
            case 2: {  // valid deposit / withdrawal (witness 40)
                VdWitness vw;
                vw.old_balance = w.sbal();
                w.opening(vw.opening_elems, vw.opening_indices);
                if (kind == 1) {
                    VdStatement st;
                    st.deposit.from = s.f();
                    st.deposit.token = s.f();
                    st.deposit.amount = s.f();
                    st.merkle_root = s.f();
                    st.old_nullifier = s.f();
                    st.new_commitment = s.f();
                    st.recovery_id = s.f();
                    st.new_amount_share = s.f();
                    vd_apply_constraints(cs, vw, st);
                } else {
                    VwStatement st;
                    st.to = s.f();
                    st.token = s.f();
                    st.amount = s.f();
                    st.merkle_root = s.f();
                    st.old_nullifier = s.f();
                    st.new_commitment = s.f();
                    st.recovery_id = s.f();
                    st.new_amount_share = s.f();
                    vw_apply_constraints(cs, vw, st);
                }
                break;
            }
            case 3: {  // valid order cancellation (witness 34)
                VocWitness vw;
                vw.old_intent = w.sint();
                w.opening(vw.opening_elems, vw.opening_idx);
                VocStatement st;
                st.merkle_root = s.f();
                st.old_intent_nullifier = s.f();
                st.owner = s.f();
                voc_apply_constraints(cs, vw, st);
                break;
            }
            case 4: {  // intent-and-balance validity (witness 91)
                ValidityWitness vw;
                vw.old_intent = w.sint();
                w.opening(vw.intent_opening_elems, vw.intent_opening_idx);
                vw.intent = w.in();
                vw.new_amount_public_share = w.f();
                vw.old_balance = w.sbal();
                w.opening(vw.balance_opening_elems, vw.balance_opening_idx);
                vw.balance = w.bal();
                vw.post_match_balance_shares = w.pms();
                ValidityStatement st;
                st.intent_merkle_root = s.f();
                st.old_intent_nullifier = s.f();
                st.intent_partial_private = s.f();
                st.intent_partial_public = s.f();
                st.intent_recovery_id = s.f();
                st.balance_merkle_root = s.f();
                st.old_balance_nullifier = s.f();
                st.balance_partial_private = s.f();
                st.balance_partial_public = s.f();
                st.balance_recovery_id = s.f();
                validity_apply_constraints(cs, vw, st, (int)L.ib_align, L.pg0, L.pg1);
                break;
            }
            case 5: {  // first-fill validity (witness 69)
                FfWitness vw;
                vw.intent = w.in();
                vw.share_stream = w.cs();
                vw.recovery_stream = w.cs();
                w.frs(vw.private_intent_shares, 5);
                vw.new_amount_public_share = w.f();
                vw.sig = w.sig();
                vw.old_balance = w.sbal();
                vw.balance = w.bal();
                vw.post_match_balance_shares = w.pms();
                w.opening(vw.opening_elems, vw.opening_idx);
                FfStatement st;
                st.merkle_root = s.f();
                s.frs(st.intent_public_share, 4);
                st.intent_private_share_commitment = s.f();
                st.intent_recovery_id = s.f();
                st.balance_partial_private = s.f();
                st.balance_partial_public = s.f();
                st.old_balance_nullifier = s.f();
                st.balance_recovery_id = s.f();
                ff_apply_constraints(cs, vw, st, (int)L.ib_align, L.pg0, L.pg1);
                break;
            }
            case 6: {  // intent-only validity (witness 39)
                IoValidityWitness vw;
                vw.old_intent = w.sint();
                w.opening(vw.opening_elems, vw.opening_idx);
                vw.intent = w.in();
                IoValidityStatement st;
                st.owner = s.f();
                st.merkle_root = s.f();
                st.old_intent_nullifier = s.f();
                st.new_amount_public_share = s.f();
                st.partial_private = s.f();
                st.partial_public = s.f();
                st.recovery_id = s.f();
                io_validity_apply_constraints(cs, vw, st, (int)L.io_align, L.io_off);
                break;
            }
            case 7: {  // intent-only first fill (witness 14)
                IoffWitness vw;
                vw.intent = w.in();
                vw.share_stream = w.cs();
                vw.recovery_stream = w.cs();
                w.frs(vw.private_shares, 5);
                IoffStatement st;
                st.owner = s.f();
                st.intent_private_commitment = s.f();
                st.recovery_id = s.f();
                s.frs(st.intent_public_share, 5);
                ioff_apply_constraints(cs, vw, st, (int)L.io_align, L.io_off);
                break;
            }
            case 8: {  // new output balance (witness 74)
                NobWitness vw;
                vw.new_balance = w.sbal();
                vw.balance = w.bal();
                vw.post_match_balance_shares = w.pms();
                vw.existing_balance = w.sbal();
                w.opening(vw.opening_elems, vw.opening_idx);
                vw.sig = w.sig();
                NobStatement st;
                st.existing_merkle_root = s.f();
                st.existing_nullifier = s.f();
                s.frs(st.pre_match_shares, 5);
                st.partial_private = s.f();
                st.partial_public = s.f();
                st.recovery_id = s.f();
                nob_apply_constraints(cs, vw, st, (int)L.ib_align, L.og0, L.og1);
                break;
            }
            case 9: {  // output-balance validity (witness 51)
                ObValidityWitness vw;
                vw.old_balance = w.sbal();
                w.opening(vw.opening_elems, vw.opening_idx);
                vw.balance = w.bal();
                vw.post_match_balance_shares = w.pms();
                ObValidityStatement st;
                st.merkle_root = s.f();
                st.old_balance_nullifier = s.f();
                st.partial_private = s.f();
                st.partial_public = s.f();
                st.recovery_id = s.f();
                ob_validity_apply_constraints(cs, vw, st, (int)L.ib_align, L.og0, L.og1);
                break;
            }
            case 11: {  // ib public settlement (witness 28)
                SettlementParty p = w.party_no_ob();
                PubSettlementStatement st;
                st.obligation = s.ob();
                st.amount_public_share = s.f();
                st.in_shares = s.pms();
                st.out_shares = s.pms();
                st.relayer_fee_repr = s.f();
                st.protocol_fee_repr = s.f();
                st.relayer_fee_recipient = s.f();
                pub_settlement_apply_constraints(cs, p, st, (int)L.ib_align, L.pg0,
                                                 L.og0);
                break;
            }
            case 12: {  // ib bounded settlement (witness 28)
                SettlementParty p = w.party_no_ob();
                IbBoundedStatement st;
                st.bmr.internal_party_input_token = s.f();
                st.bmr.internal_party_output_token = s.f();
                st.bmr.min_internal_party_amount_in = s.f();
                st.bmr.max_internal_party_amount_in = s.f();
                st.bmr.price_repr = s.f();
                st.bmr.block_deadline = s.f();
                st.amount_public_share = s.f();
                st.in_shares = s.pms();
                st.out_shares = s.pms();
                st.internal_relayer_fee_repr = s.f();
                st.external_relayer_fee_repr = s.f();
                st.relayer_fee_recipient = s.f();
                ib_bounded_apply_constraints(cs, p, st, (int)L.ib_align, L.pg0, L.og0);
                break;
            }
            case 13: {  // io public settlement (witness 5)
                Intent in = w.in();
                IoSettlementStatement st;
                st.obligation = s.ob();
                st.relayer_fee_repr = s.f();
                st.relayer_fee_recipient = s.f();
                io_settlement_apply_constraints(cs, in, st);
                break;
            }
            case 14: {  // io bounded settlement (witness 5)
                Intent in = w.in();
                IoBoundedStatement st;
                st.bmr.internal_party_input_token = s.f();
                st.bmr.internal_party_output_token = s.f();
                st.bmr.min_internal_party_amount_in = s.f();
                st.bmr.max_internal_party_amount_in = s.f();
                st.bmr.price_repr = s.f();
                st.bmr.block_deadline = s.f();
                st.internal_relayer_fee_repr = s.f();
                st.external_relayer_fee_repr = s.f();
                st.relayer_fee_recipient = s.f();
                io_bounded_apply_constraints(cs, in, st, (int)L.io_align, L.io_off);
                break;
            }
            case 15: {  // note redemption (witness 20)
                NoteRedemptionWitness vw;
                w.opening(vw.opening_elems, vw.opening_idx);
                NoteRedemptionStatement st;
                st.note = s.note();
                st.note_root = s.f();
                st.note_nullifier = s.f();
                note_redemption_apply_constraints(cs, vw, st);
                break;
            }
            case 16:
            case 17:
            case 18: {  // fee payments (witness 40 [+1 blinder for 18])
                VdWitness vw;
                vw.old_balance = w.sbal();
                w.opening(vw.opening_elems, vw.opening_indices);
                Fr blinder = (kind == 18) ? w.f() : Fr::zero();
                int field = (kind == 17) ? 6 : 5;
                int note_mode = (kind == 18) ? 1 : 0;
                std::vector<Fr> ss;
                int nst = (note_mode == 0) ? 9 : 7;
                for (int i = 0; i < nst; ++i) ss.push_back(s.f());
                fee_payment_apply_constraints(cs, vw, blinder, field, note_mode,
                                              kind != 17, ss);
                break;
            }
            case 19: {  // private protocol fee (witness 42)
                VdWitness vw;
                vw.old_balance = w.sbal();
                w.opening(vw.opening_elems, vw.opening_indices);
                Fr blinder = w.f();
                Fr enc_k = w.f();
                std::vector<Fr> ss;
                for (int i = 0; i < 14; ++i) ss.push_back(s.f());
                fee_private_protocol_apply_constraints(cs, vw, blinder, enc_k, ss);
                break;
            }
            default:
                fprintf(stderr, "rng_circ_from_scalars: unknown kind %d\n", kind);
                return nullptr;
        }
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_from_scalars(kind=%d): %s\n", kind, why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_from_scalars: %s\n", e.what());
        return nullptr;
    }
}

// VALID ORDER CANCELLATION circuit (valid_order_cancellation.rs)
void* rng_circ_build_valid_order_cancellation(uint64_t seed) {
    try {
        VocWitness w;
        VocStatement st;
        voc_build_witness_statement(seed, w, st);
        PlonkCircuit cs;
        voc_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_valid_order_cancellation: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_valid_order_cancellation: %s\n", e.what());
        return nullptr;
    }
}

// circuit builders from caller-supplied witness/statement scalars (the shape
// the external prover service receives — api_types.rs requests; Montgomery
// limbs, field order per the reference structs)
void* rng_circ_vbc_from_scalars(const uint64_t* witness12, const uint64_t* statement13) {
    try {
        VbcWitness w;
        VbcStatement st;
        vbc_witness_from_scalars((const Fr*)witness12, w);
        vbc_statement_from_scalars((const Fr*)statement13, st);
        PlonkCircuit cs;
        vbc_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_vbc_from_scalars: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_vbc_from_scalars: %s\n", e.what());
        return nullptr;
    }
}

void* rng_circ_settlement_from_scalars(const uint64_t* witness64,
                                       const uint64_t* statement17) {
    try {
        SettlementWitness w;
        SettlementStatement st;
        settlement_witness_from_scalars((const Fr*)witness64, w);
        settlement_statement_from_scalars((const Fr*)statement17, st);
        PlonkCircuit cs;
        settlement_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_settlement_from_scalars: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_settlement_from_scalars: %s\n", e.what());
        return nullptr;
    }
}

// expose the fixed-seed witness/statement generators (bench/test data)
void rng_vbc_witness_statement(uint64_t seed, uint64_t* witness12, uint64_t* statement13) {
    VbcWitness w;
    VbcStatement st;
    vbc_build_witness_statement(seed, w, st);
    auto ws = w.to_scalars();
    auto ss = st.to_scalars();
    memcpy(witness12, ws.data(), ws.size() * sizeof(Fr));
    memcpy(statement13, ss.data(), ss.size() * sizeof(Fr));
}
void rng_settlement_witness_statement(uint64_t seed, uint64_t* witness64,
                                      uint64_t* statement17) {
    SettlementWitness w;
    SettlementStatement st;
    settlement_build_witness_statement(seed, w, st);
    std::vector<Fr> ws;
    for (int i = 0; i < 2; ++i) {
        auto a = w.p[i].obligation.to_scalars();
        auto b = w.p[i].intent.to_scalars();
        auto c = w.p[i].input_balance.to_scalars();
        auto d = w.p[i].pre_in_shares.to_scalars();
        auto e = w.p[i].output_balance.to_scalars();
        auto f = w.p[i].pre_out_shares.to_scalars();
        ws.insert(ws.end(), a.begin(), a.end());
        ws.insert(ws.end(), b.begin(), b.end());
        ws.push_back(w.p[i].pre_amount_share);
        ws.insert(ws.end(), c.begin(), c.end());
        ws.insert(ws.end(), d.begin(), d.end());
        ws.insert(ws.end(), e.begin(), e.end());
        ws.insert(ws.end(), f.begin(), f.end());
    }
    auto ss = st.to_scalars();
    memcpy(witness64, ws.data(), ws.size() * sizeof(Fr));
    memcpy(statement17, ss.data(), ss.size() * sizeof(Fr));
}

// linked test circuit: `count` link-group values derived from value_seed,
// placed at (alignment, offset), plus filler gates scaled by `scale` —
// two different scales give two circuits of different domain size whose
// group values coincide (the cross-circuit linking shape: a validity proof
// linking into the settlement proof across domains).
void* rng_testcirc_build_linked(uint64_t value_seed, uint64_t scale, uint64_t alignment,
                                uint64_t offset, uint64_t count) {
    try {
        PlonkCircuit cs;
        cs.create_link_group("xlink", (int)alignment, (int64_t)offset);
        Lcg vr(value_seed);
        for (uint64_t k = 0; k < count; ++k) {
            Var v = cs.create_variable(vr.fr());
            cs.add_to_link_group(v, "xlink");
        }
        build_mixed_circuit(cs, value_seed + 1000 * scale, scale);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_testcirc_build_linked: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_testcirc_build_linked: %s\n", e.what());
        return nullptr;
    }
}

// `Valid Deposit` circuit builder (zk_circuits/valid_deposit.rs)
void* rng_circ_build_valid_deposit(uint64_t seed) {
    try {
        VdWitness w;
        VdStatement st;
        vd_build_witness_statement(seed, w, st);
        PlonkCircuit cs;
        vd_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_valid_deposit: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_valid_deposit: %s\n", e.what());
        return nullptr;
    }
}

// `Valid Withdrawal` circuit builder (zk_circuits/valid_withdrawal.rs)
void* rng_circ_build_valid_withdrawal(uint64_t seed) {
    try {
        VdWitness w;
        VwStatement st;
        vw_build_witness_statement(seed, w, st);
        PlonkCircuit cs;
        vw_apply_constraints(cs, w, st);
        std::string why;
        if (!cs.check_satisfied(&why)) {
            fprintf(stderr, "rng_circ_build_valid_withdrawal: %s\n", why.c_str());
            return nullptr;
        }
        return new CircuitTables(cs.finalize());
    } catch (const std::exception& e) {
        fprintf(stderr, "rng_circ_build_valid_withdrawal: %s\n", e.what());
        return nullptr;
    }
}

uint64_t rng_circ_num_link_groups(void* t) {
    return static_cast<CircuitTables*>(t)->link_groups.size();
}
// out: per group 3 u64 (alignment, grid offset, count), in creation order
void rng_circ_link_groups(void* t_, uint64_t* out) {
    auto* t = static_cast<CircuitTables*>(t_);
    for (size_t i = 0; i < t->link_groups.size(); ++i) {
        out[3 * i] = t->link_groups[i].alignment;
        out[3 * i + 1] = t->link_groups[i].offset;
        out[3 * i + 2] = t->link_groups[i].count;
    }
}

// native Poseidon2 hash (for cross-checks vs the oracle's restatement)
void rng_poseidon_hash(const uint64_t* inputs_mont, uint64_t n, uint64_t* out_mont) {
    std::vector<Fr> in(n);
    memcpy(in.data(), inputs_mont, n * sizeof(Fr));
    Fr r = poseidon_hash(in.data(), n);
    memcpy(out_mont, r.l, 32);
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

// field-mul microbenchmark: returns ms per launch; variant 0 = 64-bit CIOS,
// 1 = 32-bit CIOS; dep = dependent chain
double rng_bench_frmul(int variant, uint32_t blocks, uint32_t iters, int dep,
                       int reps) {
    if (!gpu_ok()) return -1;
    Fr* io = nullptr;
    if (hipMalloc(&io, 1024 * sizeof(Fr)) != hipSuccess) return -1;
    std::vector<Fr> init(1024);
    for (int i = 0; i < 1024; ++i) init[i] = Fr::from_u64(i * 2654435761u + 1);
    hipMemcpy(io, init.data(), 1024 * sizeof(Fr), hipMemcpyHostToDevice);
    auto launch = [&]() {
        if (variant == 0)
            hipLaunchKernelGGL(k_bench_frmul<0>, dim3(blocks), dim3(256), 0, 0, io,
                               iters, dep);
        else if (variant == 1)
            hipLaunchKernelGGL(k_bench_frmul<1>, dim3(blocks), dim3(256), 0, 0, io,
                               iters, dep);
        else
            hipLaunchKernelGGL(k_bench_frmul<2>, dim3(blocks), dim3(256), 0, 0, io,
                               iters, dep);
    };
    launch();
    hipDeviceSynchronize();
    hipEvent_t e0, e1;
    hipEventCreate(&e0);
    hipEventCreate(&e1);
    hipEventRecord(e0, 0);
    for (int r = 0; r < reps; ++r) launch();
    hipEventRecord(e1, 0);
    hipEventSynchronize(e1);
    float ms = 0;
    hipEventElapsedTime(&ms, e0, e1);
    hipEventDestroy(e0);
    hipEventDestroy(e1);
    hipFree(io);
    return ms / reps;
}

// pairing equality check on raw records (host; for bilinearity tests and
// external use): p1/p2 = 9-u64 G1 affine, q1/q2 = 16-u64 G2 affine.
// Returns 1 if e(p1, q1) == e(p2, q2).
int rng_pairing_check(const uint64_t* p1, const uint64_t* q1, const uint64_t* p2,
                      const uint64_t* q2) {
    if (p1[8] || p2[8]) return -1;  // infinities unsupported here
    Fq p1x, p1y, p2x, p2y;
    memcpy(p1x.l, p1, 32);
    memcpy(p1y.l, p1 + 4, 32);
    memcpy(p2x.l, p2, 32);
    memcpy(p2y.l, p2 + 4, 32);
    PG2 Q1, Q2;
    memcpy(Q1.x.a.l, q1, 32);
    memcpy(Q1.x.b.l, q1 + 4, 32);
    memcpy(Q1.y.a.l, q1 + 8, 32);
    memcpy(Q1.y.b.l, q1 + 12, 32);
    memcpy(Q2.x.a.l, q2, 32);
    memcpy(Q2.x.b.l, q2 + 4, 32);
    memcpy(Q2.y.a.l, q2 + 8, 32);
    memcpy(Q2.y.b.l, q2 + 12, 32);
    return pairing_check_eq(p1x, p1y, Q1, p2x, p2y, Q2) ? 1 : 0;
}

// host G1 addition on 9-u64 affine records (for combining per-rank MSM shard
// results after the RCCL/gloo exchange — SURVEY.md §8e; host-side, tiny)
void rng_g1_add_affine(const uint64_t* a9, const uint64_t* b9, uint64_t* out9) {
    auto load = [](const uint64_t* r) {
        G1Jac j;
        if (r[8]) return G1Jac::identity();
        G1Aff a;
        memcpy(a.x.l, r, 32);
        memcpy(a.y.l, r + 4, 32);
        return G1Jac::from_affine(a);
    };
    G1Jac s = load(a9).add(load(b9));
    jac_to_affine_record(s, out9);
}

// host keccak-256 (exported for known-answer tests of the transcript hash)
void rng_keccak256(const uint8_t* data, size_t len, uint8_t* out32) {
    keccak256_h(data, len, out32);
}

// blinder DRBG block i for a given seed (Montgomery limbs out) — lets tests
// pin the oracle/product DRBG spec equality on CPU
void rng_debug_drbg(uint64_t seed, uint32_t n_blocks, uint64_t* out) {
    HostDrbg d(seed);
    for (uint32_t i = 0; i < n_blocks; ++i) {
        Fr b = d.next();
        memcpy(out + 4 * i, b.l, 32);
    }
}

// ---- TurboPlonk prover entry points ----

struct RngProvingKey {
    PlonkPkImpl impl;
};

RngProvingKey* rng_preprocess(RngCtx* ctx, const RngCircuitDesc* desc) {
    if (!gpu_ok() || !ctx || !desc || !desc->selectors || !desc->sigma) return nullptr;
    if (desc->n < 8 || (desc->n & (desc->n - 1))) return nullptr;
    if (desc->n + 3 > ctx->impl.srs_count) return nullptr;  // SRS too small
    auto pk = std::make_unique<RngProvingKey>();
    pk->impl.ctx = &ctx->impl;
    if (plonk_preprocess_impl(&ctx->impl, desc, &pk->impl) != RNG_OK) return nullptr;
    return pk.release();
}

void rng_pk_free(RngProvingKey* pk) { delete pk; }

int rng_prove(RngCtx* ctx, const RngProvingKey* pk, const uint64_t* wires,
              const uint64_t* public_inputs, uint64_t seed, uint64_t* out_proof,
              uint64_t* out_link_hint) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!ctx || !pk || !wires || !out_proof) return RNG_ERR_BAD_ARG;
    if (pk->impl.npub > 0 && !public_inputs) return RNG_ERR_BAD_ARG;
    return plonk_prove_impl(&ctx->impl, pk->impl, (const Fr*)wires,
                            (const Fr*)public_inputs, seed, out_proof, out_link_hint);
}

int rng_prove_cohort(RngCtx* ctx, const RngProvingKey* pk, uint64_t k,
                     const uint64_t* wires, const uint64_t* public_inputs,
                     const uint64_t* seeds, uint64_t* out_proofs,
                     uint64_t* out_link_hints) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!ctx || !pk || !wires || !seeds || !out_proofs || k == 0 || k > 128)
        return RNG_ERR_BAD_ARG;
    if (pk->impl.npub > 0 && !public_inputs) return RNG_ERR_BAD_ARG;
    if (k == 1)  // same bytes by construction; the per-proof path has lower
                 // latency (fewer phase syncs) when there is nothing to fuse
        return plonk_prove_impl(&ctx->impl, pk->impl, (const Fr*)wires,
                                (const Fr*)public_inputs, seeds[0], out_proofs,
                                out_link_hints);
    return plonk_prove_cohort_impl(&ctx->impl, pk->impl, (uint32_t)k,
                                   (const Fr*)wires, (const Fr*)public_inputs, seeds,
                                   out_proofs, out_link_hints);
}

// PK introspection for tests / the verifier side
uint64_t rng_pk_n(RngProvingKey* pk) { return pk->impl.n; }
// out: 18 affine records (13 selector comms, 5 sigma comms)
void rng_pk_comms(RngProvingKey* pk, uint64_t* out) {
    auto store = [&](const G1Aff& a, bool inf, uint64_t* rec) {
        if (inf) {  // canonical infinity record: zeroed coords + flag
            memset(rec, 0, 8 * 8);
            rec[8] = 1;
            return;
        }
        memcpy(rec, a.x.l, 32);
        memcpy(rec + 4, a.y.l, 32);
        rec[8] = 0;
    };
    for (int s = 0; s < 13; ++s)
        store(pk->impl.sel_comms[s], pk->impl.sel_inf[s], out + 9 * s);
    for (int j = 0; j < 5; ++j)
        store(pk->impl.sig_comms[j], pk->impl.sig_inf[j], out + 9 * (13 + j));
}

// Full PlonK verifier with the real BN254 pairing (replaces
// PlonkKzgSnark::verify::<SolidityTranscript>, called at traits.rs:1012).
// Host-only CPU path by design: verification is the relayer-side self-check,
// not the GPU proving hot loop.
static G1Jac jac_mul_canon(const G1Jac& p, const Fr& s) {
    u64 e[4];
    s.to_canonical(e);
    G1Jac acc = G1Jac::identity();
    for (int i = 255; i >= 0; --i) {
        acc = acc.dbl();
        if ((e[i >> 6] >> (i & 63)) & 1) acc = acc.add(p);
    }
    return acc;
}

int rng_verify(RngCtx* ctx, const RngProvingKey* pkw, const uint64_t* public_inputs,
               const uint64_t* proof) {
    if (!ctx || !pkw || !proof) return RNG_ERR_BAD_ARG;
    const PlonkPkImpl& pk = pkw->impl;
    const uint64_t n = pk.n;
    if (pk.npub > 0 && !public_inputs) return RNG_ERR_BAD_ARG;
    const Fr* pubs = (const Fr*)public_inputs;
    // --- deserialize proof (layout: include/rng_prover.h) ---
    G1Jac comms[13];
    bool infs[13];
    for (int i = 0; i < 13; ++i) {
        infs[i] = proof[9 * i + 8] != 0;
        if (infs[i]) {
            comms[i] = G1Jac::identity();
        } else {
            G1Aff a;
            memcpy(a.x.l, proof + 9 * i, 32);
            memcpy(a.y.l, proof + 9 * i + 4, 32);
            comms[i] = G1Jac::from_affine(a);
        }
    }
    // Reject malformed proofs BEFORE transcript replay / pairing: the
    // reference's arkworks deserialization enforces curve membership and
    // canonical limb ranges (ark-serialize Validate::Yes on Proof fields);
    // accepting raw records here would feed off-curve points into the Miller
    // loop (invalid-point soundness hazard — ADVICE r01).
    {
        const Fq b3 = Fq::from_u64(3);  // BN254: y^2 = x^3 + 3
        for (int i = 0; i < 13; ++i) {
            if (infs[i]) continue;
            const uint64_t* rec = proof + 9 * i;
            if (Fq::geq_mod(rec) || Fq::geq_mod(rec + 4)) return RNG_ERR_VERIFY;
            Fq x, y;
            memcpy(x.l, rec, 32);
            memcpy(y.l, rec + 4, 32);
            if (!y.sqr().eq(x.sqr().mul(x).add(b3))) return RNG_ERR_VERIFY;
        }
        for (int i = 0; i < 10; ++i)
            if (Fr::geq_mod(proof + 117 + 4 * i)) return RNG_ERR_VERIFY;
    }
    Fr wire_evals[5], sigma_evals[4], z_shift_eval;
    {
        const uint64_t* e = proof + 117;
        for (int i = 0; i < 5; ++i) memcpy(wire_evals[i].l, e + 4 * i, 32);
        for (int i = 0; i < 4; ++i) memcpy(sigma_evals[i].l, e + 20 + 4 * i, 32);
        memcpy(z_shift_eval.l, e + 36, 32);
    }
    // --- transcript replay ---
    HostTranscript tr;
    h_transcript_init(tr, pk, pubs);
    auto append_comm = [&](int i) {
        G1Aff a;
        memcpy(a.x.l, proof + 9 * i, 32);
        memcpy(a.y.l, proof + 9 * i + 4, 32);
        tr.append_g1(a, infs[i]);
    };
    for (int j = 0; j < 5; ++j) append_comm(j);
    Fr beta = tr.challenge(), gamma = tr.challenge();
    append_comm(5);
    Fr alpha = tr.challenge();
    for (int i = 6; i < 11; ++i) append_comm(i);
    Fr zeta = tr.challenge();
    for (int j = 0; j < 5; ++j) tr.append_fr(wire_evals[j]);
    for (int j = 0; j < 4; ++j) tr.append_fr(sigma_evals[j]);
    tr.append_fr(z_shift_eval);
    Fr v = tr.challenge();
    append_comm(11);
    append_comm(12);
    Fr u = tr.challenge();

    Fr w = h_fr_root_of_unity((uint32_t)n);
    Fr zeta_n = zeta.pow_u64(n);
    Fr zh_zeta = zeta_n.sub(Fr::one());
    Fr l1_zeta = zh_zeta.mul(Fr::from_u64(n).mul(zeta.sub(Fr::one())).inverse());
    Fr pi_zeta = Fr::zero();
    {
        std::vector<Fr> dens(pk.npub);
        std::vector<Fr> wis(pk.npub);
        Fr wi = Fr::one();
        for (uint64_t i = 0; i < pk.npub; ++i) {
            wis[i] = wi;
            dens[i] = Fr::from_u64(n).mul(zeta.sub(wi));
            wi = wi.mul(w);
        }
        if (pk.npub) {
            dens = hbatch_inverse(dens);
            for (uint64_t i = 0; i < pk.npub; ++i)
                pi_zeta = pi_zeta.add(pubs[i].mul(wis[i]).mul(zh_zeta).mul(dens[i]));
        }
    }
    auto p5 = [](const Fr& x) {
        Fr x2 = x.sqr();
        return x2.sqr().mul(x);
    };
    const Fr* wb = wire_evals;
    Fr fbar = Fr::one(), Bbar = Fr::one();
    for (int j = 0; j < 5; ++j)
        fbar = fbar.mul(wb[j].add(beta.mul(pk.k[j]).mul(zeta)).add(gamma));
    for (int j = 0; j < 4; ++j)
        Bbar = Bbar.mul(wb[j].add(beta.mul(sigma_evals[j])).add(gamma));
    Fr ED = pi_zeta.neg()
                .add(alpha.mul(z_shift_eval).mul(Bbar).mul(wb[4].add(gamma)))
                .add(alpha.sqr().mul(l1_zeta));

    // [D]
    G1Jac D = G1Jac::identity();
    auto addsel = [&](int i, const Fr& s) {
        if (pk.sel_inf[i]) return;
        D = D.add(jac_mul_canon(G1Jac::from_affine(pk.sel_comms[i]), s));
    };
    addsel(11, Fr::one());
    for (int j = 0; j < 4; ++j) addsel(j, wb[j]);
    addsel(4, wb[0].mul(wb[1]));
    addsel(5, wb[2].mul(wb[3]));
    for (int j = 0; j < 4; ++j) addsel(6 + j, p5(wb[j]));
    addsel(12, wb[0].mul(wb[1]).mul(wb[2]).mul(wb[3]).mul(wb[4]));
    addsel(10, wb[4].neg());
    D = D.add(jac_mul_canon(comms[5], alpha.mul(fbar).add(alpha.sqr().mul(l1_zeta))));
    if (!pk.sig_inf[4])
        D = D.add(jac_mul_canon(G1Jac::from_affine(pk.sig_comms[4]),
                                alpha.mul(beta).mul(z_shift_eval).mul(Bbar).neg()));
    {
        Fr zpow = zh_zeta.neg();
        Fr step = zeta.pow_u64(n + 2);
        for (int i = 0; i < 5; ++i) {
            D = D.add(jac_mul_canon(comms[6 + i], zpow));
            zpow = zpow.mul(step);
        }
    }
    // F, E
    G1Jac F = D;
    Fr E = ED;
    Fr vp = Fr::one();
    for (int j = 0; j < 5; ++j) {
        vp = vp.mul(v);
        F = F.add(jac_mul_canon(comms[j], vp));
        E = E.add(vp.mul(wire_evals[j]));
    }
    for (int j = 0; j < 4; ++j) {
        vp = vp.mul(v);
        if (!pk.sig_inf[j])
            F = F.add(jac_mul_canon(G1Jac::from_affine(pk.sig_comms[j]), vp));
        E = E.add(vp.mul(sigma_evals[j]));
    }
    F = F.add(jac_mul_canon(comms[5], u));
    E = E.add(u.mul(z_shift_eval));

    // P1 = W + u W'; P2 = zeta W + u zeta w W' + F - E*G
    G1Jac Wz = comms[11], Wzw = comms[12];
    G1Jac P1 = Wz.add(jac_mul_canon(Wzw, u));
    G1Jac P2 = jac_mul_canon(Wz, zeta)
                   .add(jac_mul_canon(Wzw, u.mul(zeta).mul(w)))
                   .add(F);
    {
        G1Aff gen;
        static const u64 gx[4] = G1_GEN_X_MONT, gy[4] = G1_GEN_Y_MONT;
        memcpy(gen.x.l, gx, 32);
        memcpy(gen.y.l, gy, 32);
        P2 = P2.add(jac_mul_canon(G1Jac::from_affine(gen), E.neg()));
    }
    // to affine
    auto to_aff = [](const G1Jac& j, Fq& x, Fq& y) {
        Fq zi = j.Z.inverse();
        Fq zi2 = zi.sqr();
        x = j.X.mul(zi2);
        y = j.Y.mul(zi2.mul(zi));
    };
    if (P1.is_identity() || P2.is_identity()) return RNG_ERR_VERIFY;
    Fq p1x, p1y, p2x, p2y;
    to_aff(P1, p1x, p1y);
    to_aff(P2, p2x, p2y);
    // e(P1, [tau]_2) == e(P2, [1]_2)
    PG2 qh, qbh;
    memcpy(qh.x.a.l, ctx->impl.h_g2, 32);
    memcpy(qh.x.b.l, ctx->impl.h_g2 + 4, 32);
    memcpy(qh.y.a.l, ctx->impl.h_g2 + 8, 32);
    memcpy(qh.y.b.l, ctx->impl.h_g2 + 12, 32);
    memcpy(qbh.x.a.l, ctx->impl.beta_h_g2, 32);
    memcpy(qbh.x.b.l, ctx->impl.beta_h_g2 + 4, 32);
    memcpy(qbh.y.a.l, ctx->impl.beta_h_g2 + 8, 32);
    memcpy(qbh.y.b.l, ctx->impl.beta_h_g2 + 12, 32);
    return pairing_check_eq(p1x, p1y, qbh, p2x, p2y, qh) ? RNG_OK : RNG_ERR_VERIFY;
}

// Proof linking (replaces PlonkKzgSnark::link_proofs, called at
// proof_linking/intent_and_balance.rs:66-73).  Spec: oracle/plonk.hpp §link;
// hints are (n+2) wire-0 coefficients + a 9-u64 commitment record (the
// layout rng_prove emits); out = ([q], [W]) as 2 affine records (18 u64).
int rng_link_proofs(RngCtx* ctx, const RngProvingKey* pk, const uint64_t* hint_a,
                    const uint64_t* hint_b, uint64_t group_alignment,
                    uint64_t group_offset, uint64_t group_size,
                    uint64_t* out_link_proof) {
    if (!gpu_ok()) return RNG_ERR_NO_GPU;
    if (!ctx || !pk || !hint_a || !hint_b || !out_link_proof || group_size == 0)
        return RNG_ERR_BAD_ARG;
    const uint64_t n = pk->impl.n;
    const uint64_t hn = n + 2;
    if (prove_scratch_ensure(n) != RNG_OK) return RNG_ERR_HIP;
    std::vector<Fr> diff(hn);
    memcpy(diff.data(), hint_a, hn * sizeof(Fr));
    for (uint64_t i = 0; i < hn; ++i) {
        Fr b;
        memcpy(b.l, hint_b + 4 * i, 32);
        diff[i] = diff[i].sub(b);
    }
    // vanishing polynomial of S = {w_{2^a}^(off+k)} on the alignment grid
    Fr w = h_fr_root_of_unity(1u << group_alignment);
    Fr wi = w.pow_u64(group_offset);
    std::vector<Fr> zs{Fr::one()};
    for (uint64_t i = 0; i < group_size; ++i) {
        std::vector<Fr> nz(zs.size() + 1, Fr::zero());
        for (size_t j = 0; j < zs.size(); ++j) {
            nz[j + 1] = nz[j + 1].add(zs[j]);
            nz[j] = nz[j].sub(zs[j].mul(wi));
        }
        zs = std::move(nz);
        wi = wi.mul(w);
    }
    // q = diff / zs (monic long division)
    std::vector<Fr> q;
    {
        std::vector<Fr> r = diff;
        size_t dd = zs.size() - 1;
        if (r.size() < zs.size()) r.resize(zs.size(), Fr::zero());
        q.assign(r.size() - dd, Fr::zero());
        for (size_t i = r.size(); i-- > dd;) {
            Fr c = r[i];
            q[i - dd] = c;
            if (c.is_zero()) continue;
            for (size_t j = 0; j <= dd; ++j) r[i - dd + j] = r[i - dd + j].sub(zs[j].mul(c));
        }
    }
    G1Aff qc;
    bool qinf;
    if (commit_dev(&ctx->impl, q, &qc, &qinf) != RNG_OK) return RNG_ERR_HIP;
    HostTranscript tr;
    tr.append_u64(group_alignment);
    tr.append_u64(group_offset);
    tr.append_u64(group_size);
    auto append_rec = [&](const uint64_t* rec) {
        G1Aff a;
        memcpy(a.x.l, rec, 32);
        memcpy(a.y.l, rec + 4, 32);
        tr.append_g1(a, rec[8] != 0);
    };
    append_rec(hint_a + 4 * hn);
    append_rec(hint_b + 4 * hn);
    tr.append_g1(qc, qinf);
    Fr eta = tr.challenge();
    Fr zs_eta = hpoly_eval(zs, eta);
    std::vector<Fr> F = diff;
    hpoly_add_scaled(F, q, zs_eta.neg());
    std::vector<Fr> W = hpoly_div_linear(F, eta);
    G1Aff wc;
    bool winf;
    if (commit_dev(&ctx->impl, W, &wc, &winf) != RNG_OK) return RNG_ERR_HIP;
    memcpy(out_link_proof, qc.x.l, 32);
    memcpy(out_link_proof + 4, qc.y.l, 32);
    out_link_proof[8] = qinf ? 1 : 0;
    memcpy(out_link_proof + 9, wc.x.l, 32);
    memcpy(out_link_proof + 13, wc.y.l, 32);
    out_link_proof[17] = winf ? 1 : 0;
    return RNG_OK;
}

}  // extern "C"
