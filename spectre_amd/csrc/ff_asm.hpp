// ff_asm.hpp — hand-scheduled BN254 field multiply for gfx950 (CDNA4).
//
// PRODUCT CODE (device-only). The compiler's lowering of the C CIOS loop
// (ff.hpp) spends ~2 v_mov + a 64-bit add per v_mad_u64_u32 materializing
// {carry, 0} operand pairs — measured ~103 G Fq-mul/s chip-wide vs a ~172
// G/s mad-issue floor. This file removes the carry
// materialization with the one piece of hardware the C code cannot reach:
// v_mad_u64_u32's VOP3B carry-out. Each 32x32 product costs exactly
//   v_mad_u64_u32  acc, vcc, a, b, acc     (4 cycles: 64-bit mad)
//   v_addc_co_u32  ovf, vcc, 0, ovf, vcc   (2 cycles: count the carry)
// i.e. 3 issue-slots/product instead of the compiler's ~6, with column
// overflow held in a plain 32-bit counter (a column of <= 16 products can
// carry out of the 64-bit accumulator at most 16 times).
//
// Algorithm: product-scanning ("columns") Montgomery multiplication —
// low columns 0..7 are annihilated with m_k = t_k * (-p^-1) as they
// complete, high columns 8..14 emit the result limbs. Result < 2p
// (standard bound: (a*b + (<R)*p)/R < p^2/R + p < 2p), one conditional
// subtract canonicalizes. Bit-identical to ff_mul<C> by construction;
// parity enforced by tools/microbench.hip's check kernel and the whole
// GPU test suite once wired into the product kernels.
#pragma once
#include "ff.hpp"

#if defined(__HIPCC__)

// acc += a*b (64-bit), counting 64-bit carry-outs in ovf. (The asm body
// exists only in the device pass; the host pass sees a plain fallback so
// __global__ kernels type-check, and never executes it.)
__device__ __forceinline__ void ff_mad64(uint64_t& acc, uint32_t& ovf,
                                         uint32_t a, uint32_t b) {
#if defined(__HIP_DEVICE_COMPILE__)
    uint64_t c;  // explicit carry-mask pair: a vcc clobber makes the
                 // compiler pad every block boundary with s_nop hazards
    asm volatile("v_mad_u64_u32 %0, %2, %3, %4, %0\n\t"
                 "v_addc_co_u32 %1, %2, 0, %1, %2"
                 : "+v"(acc), "+v"(ovf), "=s"(c)
                 : "v"(a), "v"(b));
#else
    unsigned __int128 s = (unsigned __int128)acc + (uint64_t)a * b;
    acc = (uint64_t)s;
    ovf += (uint32_t)(s >> 64);
#endif
}

// same with a wave-uniform constant multiplicand (modulus limb -> SGPR;
// VOP3 admits one scalar source).
__device__ __forceinline__ void ff_mad64_s(uint64_t& acc, uint32_t& ovf,
                                           uint32_t a, uint32_t b_uniform) {
#if defined(__HIP_DEVICE_COMPILE__)
    uint64_t c;
    asm volatile("v_mad_u64_u32 %0, %2, %3, %4, %0\n\t"
                 "v_addc_co_u32 %1, %2, 0, %1, %2"
                 : "+v"(acc), "+v"(ovf), "=s"(c)
                 : "v"(a), "s"(b_uniform));
#else
    ff_mad64(acc, ovf, a, b_uniform);
#endif
}

template <class C>
__device__ __forceinline__ void ff_mul_cols(fp256& o, const fp256& A,
                                            const fp256& B) {
    uint32_t m[8];
    uint64_t acc = 0;
    uint32_t ovf = 0;
    // low columns: t_k + m_k * p0 == 0 mod 2^32 annihilates each column
#pragma unroll
    for (int k = 0; k < 8; k++) {
#pragma unroll
        for (int i = 0; i <= k; i++) ff_mad64(acc, ovf, A.l[i], B.l[k - i]);
#pragma unroll
        for (int i = 0; i < k; i++) ff_mad64_s(acc, ovf, m[i], C::mod(k - i));
        m[k] = (uint32_t)acc * C::inv();
        ff_mad64_s(acc, ovf, m[k], C::mod(0));
        // low word is now 0: shift down one limb, absorbing the carry count
        acc = (acc >> 32) | ((uint64_t)ovf << 32);
        ovf = 0;
    }
    // high columns emit result limbs 0..6
#pragma unroll
    for (int k = 8; k < 15; k++) {
#pragma unroll
        for (int i = k - 7; i < 8; i++) ff_mad64(acc, ovf, A.l[i], B.l[k - i]);
#pragma unroll
        for (int i = k - 7; i < 8; i++)
            ff_mad64_s(acc, ovf, m[i], C::mod(k - i));
        o.l[k - 8] = (uint32_t)acc;
        acc = (acc >> 32) | ((uint64_t)ovf << 32);
        ovf = 0;
    }
    o.l[7] = (uint32_t)acc;
    // bits >= 256 are provably zero (result < 2p < 2^255); the high half is
    // passed to the conditional subtract anyway so a violated precondition
    // reduces instead of silently truncating.
    ff_cond_sub_mod<C>(o, (uint32_t)(acc >> 32));
}

#endif  // __HIPCC__
