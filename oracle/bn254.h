/* oracle/bn254.h — CPU oracle for the Spectre/halo2 BN254 MSM + NTT hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the CPU restatement of what the
 * reference's `best_multiexp` / `best_fft` compute (see oracle/bn254.c header
 * for algorithm citations). It is used exclusively by:
 *   - tests/ (golden-vector checks and GPU-parity checks),
 *   - __graft_entry__.smoke() (as the checker), and
 *   - bench.py's `cpu_baseline` leg (timed as the reported CPU baseline).
 * The product path (libspectre_gpu.so) never links, loads or calls it.
 *
 * Parity status: pinned to the published alt_bn128/halo2curves-0.5.2
 * algorithm spec via the committed golden vectors in tests/golden/ (generated
 * by an independent Python-bigint restatement, tests/golden/generate.py).
 * Unpinned vs the reference *binary*: the reference is Rust whose arithmetic
 * lives in un-vendored crates and no Rust toolchain/network exists in this
 * environment (see DESIGN.md "Oracle pinning").
 *
 * Data formats (= halo2curves 0.5.2 memory images, little-endian):
 *   Fr/Fq element: 32 bytes = 4 x u64 LE limbs of the Montgomery residue
 *                  a*2^256 mod m.  "canonical" = 32 LE bytes of a itself.
 *   G1 affine:     64 bytes = x || y (Montgomery Fq); identity = all zeros.
 */
#ifndef SPECTRE_ORACLE_BN254_H
#define SPECTRE_ORACLE_BN254_H
#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* -- field helpers (Montgomery-form bytes in/out unless noted) ------------ */
void oracle_fr_add(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]);
void oracle_fr_sub(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]);
void oracle_fr_mul(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]);
void oracle_fr_inv(const uint8_t a[32], uint8_t out[32]);
void oracle_fr_pow(const uint8_t a[32], const uint8_t e_canon[32], uint8_t out[32]);
void oracle_fr_to_canonical(const uint8_t a[32], uint8_t out[32]);
void oracle_fr_from_canonical(const uint8_t a[32], uint8_t out[32]);
void oracle_fq_add(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]);
void oracle_fq_sub(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]);
void oracle_fq_mul(const uint8_t a[32], const uint8_t b[32], uint8_t out[32]);
void oracle_fq_inv(const uint8_t a[32], uint8_t out[32]);

/* -- G1 ------------------------------------------------------------------- */
void oracle_g1_add(const uint8_t a[64], const uint8_t b[64], uint8_t out[64]);
void oracle_g1_neg(const uint8_t a[64], uint8_t out[64]);
void oracle_g1_mul(const uint8_t p[64], const uint8_t k_canon[32], uint8_t out[64]);
int  oracle_g1_is_on_curve(const uint8_t p[64]);

/* G2 (Fq2 twist; SURVEY §8a minor row — size <= 2, oracle-only, no kernel).
 * Affine image: x.c0||x.c1||y.c0||y.c1 (32 B LE Montgomery each);
 * identity = 128 zero bytes. Scalars canonical LE. */
void oracle_g2_add(const uint8_t a[128], const uint8_t b[128], uint8_t out[128]);
void oracle_g2_neg(const uint8_t a[128], uint8_t out[128]);
void oracle_g2_mul(const uint8_t p[128], const uint8_t k_canon[32], uint8_t out[128]);
int  oracle_g2_is_on_curve(const uint8_t p[128]);
void oracle_msm_g2(const uint8_t* bases, const uint8_t* scalars_canon,
                   uint64_t n, uint8_t out[128]);

/* -- the hot path --------------------------------------------------------- */
/* Sum_i scalars[i]*bases[i]; scalars 32B each (canonical if
 * scalars_canonical, else Montgomery); bases 64B affine; OpenMP Pippenger. */
void oracle_msm_g1(const uint8_t* bases, const uint8_t* scalars, uint64_t n,
                   int scalars_canonical, uint8_t out[64]);
/* In-place radix-2 NTT over Fr, Montgomery-form data, semantics of
 * halo2 best_fft/EvaluationDomain (DESIGN.md "NTT semantics"):
 *   if coset_gen && !inverse:  data[i] *= coset_gen^i   (before)
 *   data <- DFT(data, omega)   [caller passes omega_inv for the inverse]
 *   if inverse:                data[i] *= n^{-1}
 *   if coset_gen && inverse:   data[i] *= coset_gen^i   (after; pass g^{-1})
 */
void oracle_ntt_fr(uint8_t* data, uint32_t log_n, const uint8_t omega[32],
                   int inverse, const uint8_t* coset_gen);

/* -- deterministic input generation (contract: tests/golden/generate.py) -- */
void oracle_gen_msm_inputs(uint64_t n, uint64_t seed,
                           uint8_t* scalars_canon, uint8_t* bases);
/* bench-scale variant: same scalar stream; bases B_0 = fr()*G, B_{i+1} = B_i + G */
void oracle_gen_msm_inputs_fast(uint64_t n, uint64_t seed,
                                uint8_t* scalars_canon, uint8_t* bases);
void oracle_gen_fr_vector(uint64_t n, uint64_t seed, uint8_t* out_mont);

int oracle_num_threads(void);

#ifdef __cplusplus
}
#endif
#endif
