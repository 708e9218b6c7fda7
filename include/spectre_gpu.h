/* spectre_gpu.h — C ABI of the MI355X-native BN254 MSM/NTT proving backend.
 *
 * This is the drop-in boundary for the Spectre/halo2 proving hot path: the
 * reference's own seam is the pair of free functions
 *   best_multiexp(coeffs: &[Fr], bases: &[G1Affine]) -> G1   and
 *   best_fft(a: &mut [Fr], omega: Fr, log_n: u32)
 * in halo2_proofs::arithmetic / halo2curves (reached from every proof and
 * keygen via lightclient-circuits/src/util/circuit.rs:131,158,177,211 in the
 * reference tree). A patched halo2_proofs calls these entry points through a
 * ~100-line unsafe extern "C" block; see INTEGRATION.md for the Rust-side
 * binding a maintainer would add.
 *
 * Data formats (= halo2curves-axiom 0.5.2 memory images, little-endian):
 *   Fr/Fq element : 32 B = 4 x u64 LE limbs of the Montgomery residue
 *                   a*2^256 mod m (the raw &[Fr] slice memory).
 *   canonical Fr  : 32 LE bytes of the integer itself (Fr::to_repr()).
 *   G1 affine     : 64 B = x || y Montgomery Fq; identity = 64 zero bytes.
 *   G1 Jacobian   : 96 B = X || Y || Z Montgomery Fq; identity: Z = 0
 *                   (shard partial sums only; never a caller-visible result).
 *
 * Thread safety: all calls on one ctx are serialized internally; halo2 may
 * invoke commits from concurrent rayon workers. Errors: 0 = ok, negative =
 * failure; spectre_gpu_last_error() returns a thread-local message. The
 * library REQUIRES a visible AMD GPU for every compute entry point (there is
 * deliberately no CPU fallback — a missing GPU fails loudly); the only
 * host-only entry points are spectre_gpu_msm_g1_combine and the
 * format/version queries.
 */
#ifndef SPECTRE_GPU_H
#define SPECTRE_GPU_H
#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct spectre_gpu_ctx spectre_gpu_ctx;

/* scalars argument encoding for MSM entry points */
#define SPECTRE_SCALARS_MONTGOMERY 0u /* raw &[Fr] memory (the real seam) */
#define SPECTRE_SCALARS_CANONICAL 1u  /* already to_repr()'d */

#define SPECTRE_MSM_WINDOW_BITS 16u /* signed window width c */
#define SPECTRE_MSM_NUM_WINDOWS 16u /* ceil(255/c) windows */
#define SPECTRE_MSM_MAX_BATCH 32    /* max scalar vectors per batched call */

/* ctx over `device_count` HIP devices (device_ids NULL -> 0..count-1).
 * Returns NULL on failure (no GPU, bad ids). */
spectre_gpu_ctx* spectre_gpu_init(int device_count, const int* device_ids);
void spectre_gpu_destroy(spectre_gpu_ctx*);
const char* spectre_gpu_last_error(void); /* thread-local, valid until next call */
int spectre_gpu_device_count(spectre_gpu_ctx*);
const char* spectre_gpu_version(void);

/* ---- MSM (host pointers) ------------------------------------------------
 * out_affine = Sum_i scalars[i] * bases[i].  bases_id != 0: the uploaded
 * bases are cached on-device under (bases_id, n) and `bases` may be NULL on
 * subsequent calls (SRS reuse across proofs). num_gpus in [1, device_count]:
 * the scalar/base stream is sharded contiguously, each device runs the full
 * Pippenger pipeline on its shard, partial window sums are combined on host
 * in rank order. */
int spectre_gpu_msm_g1(spectre_gpu_ctx*, uint64_t bases_id,
                       const uint8_t* bases /* n*64B */,
                       const uint8_t* scalars /* n*32B */, uint64_t n,
                       uint32_t flags, int num_gpus, uint8_t out_affine[64]);

/* Batched MSM: `nbatch` scalar vectors (batch-major, nbatch*n*32 B) over ONE
 * shared base set — the shape of create_proof's back-to-back column commits
 * (same SRS); one fused sort/accumulate pass, bases stay cache-resident
 * across the batch. out_affine receives nbatch results (64 B each). */
int spectre_gpu_msm_g1_batch(spectre_gpu_ctx*, uint64_t bases_id,
                             const uint8_t* bases /* n*64B */,
                             const uint8_t* scalars /* nbatch*n*32B */,
                             uint32_t nbatch, uint64_t n, uint32_t flags,
                             uint8_t* out_affine);
int spectre_gpu_msm_g1_batch_device(spectre_gpu_ctx*, int dev,
                                    const void* d_bases,
                                    const void* d_scalars, uint32_t nbatch,
                                    uint64_t n, uint32_t flags,
                                    uint8_t* out_affine);

/* ---- MSM (device-resident, single device `dev`) ------------------------ */
int spectre_gpu_msm_g1_device(spectre_gpu_ctx*, int dev,
                              const void* d_bases, const void* d_scalars,
                              uint64_t n, uint32_t flags,
                              uint8_t out_affine[64]);
/* Per-window Jacobian partial sums of this shard (for cross-process
 * sharding, e.g. one rank per GPU exchanging partials over RCCL):
 * out_partials = SPECTRE_MSM_NUM_WINDOWS * 96 bytes. */
int spectre_gpu_msm_g1_shard_device(spectre_gpu_ctx*, int dev,
                                    const void* d_bases,
                                    const void* d_scalars, uint64_t n,
                                    uint32_t flags, uint8_t* out_partials);
/* As shard_device, plus per-stage timings from HIP events recorded on the
 * library's own stream (the bench's live roofline measurement):
 * out_ms[0..5] = digits, radix-sort, offsets, bucket-accumulate, chunks,
 * reduce-tree; out_ms[6] = total GPU ms; out_ms[7] = count of real
 * (non-zero-digit) sort entries processed. */
int spectre_gpu_msm_g1_shard_device_timed(spectre_gpu_ctx*, int dev,
                                          const void* d_bases,
                                          const void* d_scalars, uint64_t n,
                                          uint32_t flags,
                                          uint8_t* out_partials,
                                          double out_ms[8]);
/* Pipelined (async) shard MSM: enqueue the full pipeline on one of two
 * per-device slots (own stream + scratch) and return WITHOUT synchronizing;
 * *out_slot receives the slot id. Back-to-back MSMs on alternating slots
 * overlap call i's latency-bound reduction tail with call i+1's
 * digits/sort/accumulate (create_proof's ~45 commits per proof arrive
 * exactly this way). out_partials (NUM_WINDOWS * 96 B) must stay valid
 * until spectre_gpu_msm_slot_wait(ctx, dev, slot) returns. At most one
 * call may be in flight per slot. */
int spectre_gpu_msm_g1_shard_device_async(spectre_gpu_ctx*, int dev,
                                          const void* d_bases,
                                          const void* d_scalars, uint64_t n,
                                          uint32_t flags,
                                          uint8_t* out_partials,
                                          int* out_slot);
int spectre_gpu_msm_slot_wait(spectre_gpu_ctx*, int dev, int slot);
/* Combine gathered shard partials (host-only, deterministic rank order):
 * partials = nshards * NUM_WINDOWS * 96 B. */
int spectre_gpu_msm_g1_combine(const uint8_t* partials, uint32_t nshards,
                               uint8_t out_affine[64]);

/* ---- window-sharded multi-GPU --------------------------------------------
 * Scalar-chunk sharding (above) divides only the bucket work: the
 * per-window reduction tail is fixed per rank, so strong scaling stalls.
 * Window sharding instead gives rank r of R the windows
 * [r*NUM_WINDOWS/R, (r+1)*NUM_WINDOWS/R) over ALL n points: bucket work AND
 * tail divide by R, and the exchange is a pure allgather of DISJOINT window
 * sums (no reduction at all). R must divide NUM_WINDOWS (1,2,4,8,16).
 * out_partials = w_cnt * 96 B Jacobian sums. */
int spectre_gpu_msm_g1_shard_windows_device(spectre_gpu_ctx*, int dev,
                                            const void* d_bases,
                                            const void* d_scalars, uint64_t n,
                                            uint32_t flags, uint32_t w_lo,
                                            uint32_t w_cnt,
                                            uint8_t* out_partials);
/* async variant on the pipeline slots (same contract as
 * spectre_gpu_msm_g1_shard_device_async). */
int spectre_gpu_msm_g1_shard_windows_device_async(
    spectre_gpu_ctx*, int dev, const void* d_bases, const void* d_scalars,
    uint64_t n, uint32_t flags, uint32_t w_lo, uint32_t w_cnt,
    uint8_t* out_partials, int* out_slot);
/* Concatenated rank-ordered slices (nshards * (NUM_WINDOWS/nshards) * 96 B)
 * -> affine result. nshards must divide NUM_WINDOWS. */
int spectre_gpu_msm_g1_combine_windows(const uint8_t* partials,
                                       uint32_t nshards,
                                       uint8_t out_affine[64]);

/* ---- NTT ----------------------------------------------------------------
 * In-place radix-2 NTT over Fr, Montgomery-form data, exactly halo2's
 * best_fft / EvaluationDomain semantics:
 *   if coset_gen && !inverse : data[i] *= coset_gen^i            (before)
 *   data <- DFT(data, omega)     [caller passes omega_inv to invert]
 *   if inverse               : data[i] *= n^{-1}
 *   if coset_gen && inverse  : data[i] *= coset_gen^i   (after; pass g^{-1})
 * log_n <= 28 (= BN254 Fr's 2-adicity; the reference needs 2^20..2^24 for
 * its circuits and 2^25..2^26 extended domains for aggregation). */
int spectre_gpu_ntt_fr(spectre_gpu_ctx*, uint8_t* data, uint32_t log_n,
                       const uint8_t omega[32], int inverse,
                       const uint8_t* coset_gen);
int spectre_gpu_ntt_fr_device(spectre_gpu_ctx*, int dev, void* d_data,
                              uint32_t log_n, const uint8_t omega[32],
                              int inverse, const uint8_t* coset_gen);

/* ---- pointwise Fr vector ops (device buffers, Montgomery form) ----------
 * The quotient phase evaluates gate expressions pointwise between the
 * iFFT and coset-FFT passes; these ops let a patched evaluator keep
 * polynomials device-resident. out may alias a or b.
 *   op 0: out = a + b        op 1: out = a - b       op 2: out = a * b
 *   op 3: out = a * c        op 4: out = a + c * b   (c = one Fr element)
 */
/* ---- gate-expression evaluator (quotient phase) -------------------------
 * Evaluates one custom-gate expression over every row of a (usually
 * extended-domain) column set in a single launch — the device-resident
 * iFFT -> gate-eval -> coset-FFT quotient pipeline needs O(1) launches per
 * gate instead of per-op round trips.
 *
 * program = nops * 3 uint32 words {op, a, b}, a stack machine:
 *   COL   a=column index, b=(int32) rotation: push cols[a][(row + b*rot_scale) mod n]
 *   CONST a=constant index: push constants[a]  (challenges, coefficients)
 *   ADD/SUB/MUL: pop two (second-from-top OP top), push result
 *   NEG: negate top
 * Stack depth is validated <= SPECTRE_GATE_MAX_DEPTH. All values are
 * Montgomery Fr (32 B LE). On exit, for each row:
 *   y == NULL : out[row]  = result               (first gate)
 *   y != NULL : out[row] = out[row]*y + result   (halo2's h = h*y + gate)
 * d_cols is a HOST array of ncols DEVICE pointers; constants/program/y are
 * host pointers. */
#define SPECTRE_GATE_OP_COL 0
#define SPECTRE_GATE_OP_CONST 1
#define SPECTRE_GATE_OP_ADD 2
#define SPECTRE_GATE_OP_SUB 3
#define SPECTRE_GATE_OP_MUL 4
#define SPECTRE_GATE_OP_NEG 5
#define SPECTRE_GATE_MAX_DEPTH 8
int spectre_gpu_fr_gate_eval(spectre_gpu_ctx*, int dev,
                             const void* const* d_cols, uint32_t ncols,
                             const uint8_t* constants, uint32_t nconst,
                             const uint32_t* program, uint32_t nops,
                             uint64_t n, uint32_t rot_scale,
                             const uint8_t* y /* 32 B or NULL */,
                             void* d_out);

#define SPECTRE_VEC_ADD 0
#define SPECTRE_VEC_SUB 1
#define SPECTRE_VEC_MUL 2
#define SPECTRE_VEC_SCALE 3
#define SPECTRE_VEC_ADD_SCALED 4
int spectre_gpu_fr_vec_op(spectre_gpu_ctx*, int dev, int op,
                          const void* d_a, const void* d_b /* NULL for op 3 */,
                          const uint8_t c[32] /* NULL for ops 0-2 */,
                          void* d_out, uint64_t n);

/* ---- device memory helpers (for C callers and the bench harness) ------- */
int spectre_gpu_malloc(spectre_gpu_ctx*, int dev, size_t bytes, void** d_ptr);
int spectre_gpu_free(spectre_gpu_ctx*, int dev, void* d_ptr);
int spectre_gpu_upload(spectre_gpu_ctx*, int dev, void* d_dst, const void* src,
                       size_t bytes);
int spectre_gpu_download(spectre_gpu_ctx*, int dev, void* dst,
                         const void* d_src, size_t bytes);
int spectre_gpu_synchronize(spectre_gpu_ctx*, int dev);

#ifdef __cplusplus
}
#endif
#endif /* SPECTRE_GPU_H */
