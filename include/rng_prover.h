/* renegade_amd — MI355X-native PlonK prover backend: C ABI.
 *
 * This is the drop-in boundary of the replacement prover.
 * Each entry point mirrors an extern call the reference's circuits crate
 * makes today (citations into /root/reference):
 *
 *   rng_prover_init      <- SRS load: parse_ptau_file semantics of
 *                           crates/circuits/circuit-types/src/primitives/srs.rs:63-214
 *                           (SYSTEM_SRS lazy static, srs.rs:30-42)
 *   rng_preprocess       <- PlonkKzgSnark::preprocess(&SYSTEM_SRS, &cs),
 *                           called at circuit-types/src/traits.rs:850
 *   rng_prove            <- PlonkKzgSnark::prove_with_link_hint::<_,_,SolidityTranscript>,
 *                           called at circuit-types/src/traits.rs:996
 *   rng_link_proofs      <- PlonkKzgSnark::link_proofs::<SolidityTranscript>,
 *                           called at circuits-core/src/zk_circuits/proof_linking/
 *                           intent_and_balance.rs:66-73
 *   rng_verify           <- PlonkKzgSnark::verify::<SolidityTranscript>,
 *                           called at circuit-types/src/traits.rs:1012 (CPU pairing)
 *   rng_msm_g1 / rng_ntt_fr <- the MSM/NTT primitives the north star names
 *                           (arkworks ark-ec Pippenger / ark-poly radix-2 FFT,
 *                           non-vendored deps)
 *
 * Conventions (pinned by the reference's rkyv shims,
 * types-proofs/src/rkyv_impls/plonk_proof_def.rs):
 *   - Fr/Fq elements: 4 x u64 little-endian Montgomery limbs (R = 2^256),
 *     except scalars marked `canonical` (plain little-endian integers).
 *   - G1 affine: x(4 u64), y(4 u64), infinity(u64 0/1)  => 9 u64 = 72 B.
 *   - Proof buffer layout = the rkyv field order (plonk_proof_def.rs:197-222):
 *     5 wire comms, 1 perm comm, 5 split-quotient comms, opening proof,
 *     shifted opening proof (13 G1 affine records = 117 u64), then
 *     evals: 5 wire evals, 4 sigma evals, 1 perm-next eval (10 Fr = 40 u64).
 *     Total RNG_PROOF_U64S = 157 u64.
 *   - Status codes map onto ProverError/PlonkError
 *     (circuit-types/src/errors.rs): 0 ok, negative = error below.
 *   - Thread-safety: context and pk handles are read-only after creation and
 *     may be shared across threads (the reference proves from a rayon pool,
 *     native_proof_manager.rs:143-148); each concurrent rng_prove call uses
 *     its own HIP stream.
 *
 * A GPU box is REQUIRED for the compute entry points: they fail loudly
 * (RNG_ERR_NO_GPU) rather than fall back to any CPU path.
 */
#ifndef RNG_PROVER_H
#define RNG_PROVER_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define RNG_OK 0
#define RNG_ERR_NO_GPU (-1)
#define RNG_ERR_BAD_ARG (-2)
#define RNG_ERR_HIP (-3)
#define RNG_ERR_SRS (-4)          /* maps IoError from parse_ptau_file */
#define RNG_ERR_UNSATISFIED (-5)  /* maps PlonkError::WrongProof / circuit unsat */
#define RNG_ERR_VERIFY (-6)

#define RNG_PROOF_G1S 13
#define RNG_PROOF_EVALS 10
#define RNG_PROOF_U64S (RNG_PROOF_G1S * 9 + RNG_PROOF_EVALS * 4) /* 157 */

typedef struct RngCtx RngCtx;
typedef struct RngProvingKey RngProvingKey;

/* ---- lifecycle ---- */

/* Parse a snarkjs-layout .ptau byte buffer (srs.rs:63-214 semantics) and
 * upload the G1 powers to the GPU. max_degree + 1 G1 points are read. */
RngCtx* rng_prover_init(const uint8_t* srs_ptau, size_t len, uint64_t max_degree);
void rng_ctx_free(RngCtx* ctx);

/* Deterministic TEST SRS in the snarkjs subset layout parse_ptau_file reads
 * (srs.rs:63-214): tau = keccak256("renegade-amd-srs-tau" || le64(seed)) mod r.
 * Dev/test only — production passes real ceremony bytes to rng_prover_init.
 * rng_srs_test_ptau_size returns the byte size for `power` (0 = bad power);
 * rng_srs_gen_test_ptau fills a caller buffer of at least that size. */
uint64_t rng_srs_test_ptau_size(int power);
int rng_srs_gen_test_ptau(int power, uint64_t seed, uint8_t* out, size_t out_len);

/* Library/device introspection */
int rng_gpu_available(void);
const char* rng_version(void);
/* Join the library's host worker-pool threads (profiler-finalizer hygiene;
 * pool work runs serially afterwards).  Safe to call at any time. */
void rng_shutdown_pool(void);
int rng_set_device(int device);

/* Per-kernel HIP-event times (ms) of the LAST rng_msm_* / rng_ntt_* call on
 * this thread, for roofline accounting: msm: [digits, sort, bucket_reduce,
 * window_chunks, final]; ntt: [pass1(col/small), pass2(row)].  Returns number
 * of entries written. */
int rng_msm_last_times(double out_ms[5]);
int rng_ntt_last_times(double out_ms[2]);

/* ---- primitives (the two flagship kernels) ---- */

/* In-place radix-2 NTT over BN254 Fr, natural order in/out,
 * data = batch * n elements of 4 u64 Montgomery limbs (host buffer).
 * n power of two, 2 <= n <= 2^26. */
int rng_ntt_fr(RngCtx* ctx, uint64_t* data, uint64_t n, uint64_t batch, int inverse);

/* MSM over BN254 G1: bases = n * 8 u64 (affine x,y Montgomery; no points at
 * infinity), scalars = n * 4 u64 CANONICAL little-endian. out = 9 u64 affine
 * record. window_c in [8,16], 0 = auto. */
int rng_msm_g1(RngCtx* ctx, const uint64_t* bases, const uint64_t* scalars,
               uint64_t n, uint64_t* out9, int window_c);

/* Device-resident variants for benchmarking/pipelining (inputs already in
 * HBM; see rng_dbuf_*). */
int rng_ntt_fr_dev(RngCtx* ctx, void* dev_data, uint64_t n, uint64_t batch, int inverse);
/* out-of-place: result lands in dev_out (dev_in preserved for n > 1024;
 * for n <= 1024 the single-workgroup path runs in dev_in then copies). */
int rng_ntt_fr_dev_oop(RngCtx* ctx, void* dev_in, void* dev_out, uint64_t n,
                       uint64_t batch, int inverse);
int rng_msm_g1_dev(RngCtx* ctx, const void* dev_bases, const void* dev_scalars,
                   uint64_t n, uint64_t* out9, int window_c);

/* Device buffer helpers */
void* rng_dbuf_alloc(size_t bytes);
void rng_dbuf_free(void* dbuf);
int rng_dbuf_upload(void* dbuf, const void* host_src, size_t bytes);
int rng_dbuf_download(const void* dbuf, void* host_dst, size_t bytes);
int rng_device_sync(void);

/* The SRS G1 powers already resident in HBM (bases for KZG commitments):
 * returns device pointer (n*8 u64 packed affine) and count. */
const void* rng_srs_dev_bases(RngCtx* ctx, uint64_t* count);

/* ---- PlonK prover (see DESIGN.md for the round structure) ---- */

/* Circuit description: the output of arithmetization, i.e. exactly what
 * PlonkKzgSnark::preprocess consumes from a finalized PlonkCircuit
 * (traits.rs:832-855): n gates (padded to a power of two), 13 selector
 * columns, the wire permutation over 5*n wire slots, and public-input count.
 * Layout documented in DESIGN.md §ABI. */
typedef struct {
    uint64_t n;             /* evaluation domain size (power of two) */
    uint64_t num_public;    /* number of public inputs */
    const uint64_t* selectors;  /* 13 * n Fr (Montgomery), column-major q[sel][gate] */
    const uint64_t* sigma;      /* 5 * n u64 indices: wire permutation (values < 5n) */
    uint64_t num_link_groups;
    const uint64_t* link_offsets;  /* per group: (offset, stride=1?, count) triples */
} RngCircuitDesc;

RngProvingKey* rng_preprocess(RngCtx* ctx, const RngCircuitDesc* desc);
void rng_pk_free(RngProvingKey* pk);

/* witness: 5*n Fr wire values (Montgomery, column-major w[wire][gate]);
 * public inputs: num_public Fr. seed drives the blinder (the reference uses
 * thread_rng at traits.rs:994; a fixed seed pins proof bytes for parity).
 * out_proof: RNG_PROOF_U64S u64. out_link_hint: optional (NULL to skip),
 * n+? Fr coefficients + 9 u64 commitment; see DESIGN.md. */
int rng_prove(RngCtx* ctx, const RngProvingKey* pk, const uint64_t* wires,
              const uint64_t* public_inputs, uint64_t seed,
              uint64_t* out_proof, uint64_t* out_link_hint);

/* Cohort prover: k proofs of the SAME circuit advanced in lockstep, each
 * round's commitments fused into one GPU MSM run — the batched shape of the
 * reference's proof-job pool (native_proof_manager.rs:143-148,193-198).
 * wires = k consecutive 5n-scalar wire blocks, public_inputs = k consecutive
 * statement blocks, seeds = k blinder seeds; out_proofs = k consecutive
 * 157-u64 proofs; out_link_hints (optional) = k consecutive hint buffers.
 * Proof p is bit-identical to rng_prove with seeds[p].  1 <= k <= 128. */
int rng_prove_cohort(RngCtx* ctx, const RngProvingKey* pk, uint64_t k,
                     const uint64_t* wires, const uint64_t* public_inputs,
                     const uint64_t* seeds, uint64_t* out_proofs,
                     uint64_t* out_link_hints);

int rng_verify(RngCtx* ctx, const RngProvingKey* pk, const uint64_t* public_inputs,
               const uint64_t* proof);

/* group placement per mpc-relation GroupLayout: values live on the shared
 * subgroup H_{2^alignment} at grid offsets [offset, offset+size) — the
 * placement rng_circ_link_groups reports. */
int rng_link_proofs(RngCtx* ctx, const RngProvingKey* pk, const uint64_t* hint_a,
                    const uint64_t* hint_b, uint64_t group_alignment,
                    uint64_t group_offset, uint64_t group_size,
                    uint64_t* out_link_proof);

/* ---- circuit builders (arithmetization of the reference's 20 circuits;
 *      prover_service_client.rs:100-147 route bodies) ----
 *
 * rng_circ_from_scalars(kind, witness64, statement64) builds any circuit
 * from flat scalar arrays in the reference struct field order (Montgomery
 * limbs; CSPRNG indexes and Merkle-path bits as canonical integers).
 * Returns an opaque CircuitTables handle, or NULL if the witness does not
 * satisfy the circuit.  Scalar counts per kind via rng_ws_sizes; fixed-seed
 * test vectors via rng_witness_statement.
 *
 * kind  circuit (reference file under circuits-core/src/zk_circuits/)
 *   1   VALID DEPOSIT                       valid_deposit.rs
 *   2   VALID WITHDRAWAL                    valid_withdrawal.rs
 *   3   VALID ORDER CANCELLATION            valid_order_cancellation.rs
 *   4   INTENT AND BALANCE VALIDITY         validity_proofs/intent_and_balance.rs
 *   5   ... FIRST FILL VALIDITY             validity_proofs/intent_and_balance_first_fill.rs
 *   6   INTENT ONLY VALIDITY                validity_proofs/intent_only.rs
 *   7   INTENT ONLY FIRST FILL VALIDITY     validity_proofs/intent_only_first_fill.rs
 *   8   NEW OUTPUT BALANCE VALIDITY         validity_proofs/new_output_balance.rs
 *   9   OUTPUT BALANCE VALIDITY             validity_proofs/output_balance.rs
 *  11   INTENT AND BALANCE PUBLIC SETTLEMENT   settlement/intent_and_balance_public_settlement.rs
 *  12   INTENT AND BALANCE BOUNDED SETTLEMENT  settlement/intent_and_balance_bounded_settlement.rs
 *  13   INTENT ONLY PUBLIC SETTLEMENT       settlement/intent_only_public_settlement.rs
 *  14   INTENT ONLY BOUNDED SETTLEMENT      settlement/intent_only_bounded_settlement.rs
 *  15   VALID NOTE REDEMPTION               fees/valid_note_redemption.rs
 *  16   VALID PUBLIC RELAYER FEE PAYMENT    fees/valid_public_relayer_fee_payment.rs
 *  17   VALID PUBLIC PROTOCOL FEE PAYMENT   fees/valid_public_protocol_fee_payment.rs
 *  18   VALID PRIVATE RELAYER FEE PAYMENT   fees/valid_private_relayer_fee_payment.rs
 *  19   VALID PRIVATE PROTOCOL FEE PAYMENT  fees/valid_private_protocol_fee_payment.rs
 * (VALID BALANCE CREATE and the private settlement keep their dedicated
 *  entry points rng_circ_vbc_from_scalars / rng_circ_settlement_from_scalars.)
 */
void* rng_circ_from_scalars(int kind, const uint64_t* witness64,
                            const uint64_t* statement64);
int rng_ws_sizes(int kind, uint64_t* out_num_witness, uint64_t* out_num_statement);
int rng_witness_statement(int kind, uint64_t seed, uint64_t* out_witness,
                          uint64_t* out_statement);
/* Party-aware vectors for the per-party validity kinds 4 and 9: party 0/1 of
 * the seed's consistent production bundle (the private-settlement route needs
 * all four external hints — native_proof_manager.rs:554-590). */
int rng_witness_statement_party(int kind, uint64_t seed, uint64_t party,
                                uint64_t* out_witness, uint64_t* out_statement);

/* Fixed-seed circuit builders (deterministic witness/statement generators
 * mirroring the reference's test_helpers; returns an opaque CircuitTables
 * handle or NULL if unsatisfied).  Two-arg builders take (seed, party). */
void* rng_circ_build_vbc(uint64_t seed);
void* rng_circ_build_settlement(uint64_t seed);
void* rng_circ_build_settlement_bundle(uint64_t seed);
void* rng_circ_build_valid_deposit(uint64_t seed);
void* rng_circ_build_valid_withdrawal(uint64_t seed);
void* rng_circ_build_valid_order_cancellation(uint64_t seed);
void* rng_circ_build_validity(uint64_t seed, uint64_t party);
void* rng_circ_build_ob_validity(uint64_t seed, uint64_t party);
void* rng_circ_build_ff_validity(uint64_t seed, uint64_t party);
void* rng_circ_build_io_validity(uint64_t seed);
void* rng_circ_build_ioff(uint64_t seed);
void* rng_circ_build_nob_validity(uint64_t seed);
void* rng_circ_build_public_settlement(uint64_t seed);
void* rng_circ_build_ib_bounded_settlement(uint64_t seed);
void* rng_circ_build_io_settlement(uint64_t seed);
void* rng_circ_build_io_bounded_settlement(uint64_t seed);
void* rng_circ_build_note_redemption(uint64_t seed);
void* rng_circ_build_fee_public_relayer(uint64_t seed);
void* rng_circ_build_fee_public_protocol(uint64_t seed);
void* rng_circ_build_fee_private_relayer(uint64_t seed);
void* rng_circ_build_fee_private_protocol(uint64_t seed);

/* CircuitTables accessors (the finalized arithmetization rng_preprocess /
 * rng_prove consume; column layouts as in RngCircuitDesc) */
uint64_t rng_circ_n(void* tables);
uint64_t rng_circ_npub(void* tables);
void rng_circ_get(void* tables, uint64_t* selectors, uint64_t* sigma,
                  uint64_t* wires, uint64_t* public_inputs);
uint64_t rng_circ_num_link_groups(void* tables);
/* per group: (alignment, grid offset, count) triples */
void rng_circ_link_groups(void* tables, uint64_t* out3xN);
void rng_circ_free(void* tables);

#ifdef __cplusplus
}
#endif
#endif /* RNG_PROVER_H */
