// ff.hpp — BN254 prime-field arithmetic, 8x32-bit limbs, Montgomery form.
//
// PRODUCT CODE (part of libspectre_gpu.so). Compiles both as HIP device code
// (gfx950) and as plain host C++ (the FFI's final-reduction path). 32-bit
// limbs are chosen for the CDNA4 VALU: a 256-bit Montgomery multiply (CIOS)
// lowers to chains of 32x32->64 multiply-adds (v_mad_u64_u32 class), which is
// the native integer-multiply shape on gfx950. This is deliberately a
// DIFFERENT limb decomposition from the CPU oracle's 4x64/__int128 so the two
// implementations cannot share a limb-level bug.
//
// Memory format (= halo2curves-axiom 0.5.2 memory image, the reference's
// arithmetic dependency — /root/reference/Cargo.toml:53): a field element is
// 32 little-endian bytes of the Montgomery residue a*2^256 mod m, i.e. the
// limb array IS the byte image on a little-endian machine (both host and
// gfx950 are LE). "Canonical" = 32 LE bytes of a itself (Fr::to_repr()).
#pragma once
#include <stdint.h>
#include <string.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#include <hip/hip_runtime.h>
#define FF_HD __host__ __device__ __forceinline__
#else
#define FF_HD inline
#endif

struct fp256 {
    uint32_t l[8];
};

// ---- constant providers -----------------------------------------------------
// Constants derived from the published alt_bn128 moduli (values verified
// against tests/golden fixtures; see tests/golden/bn254_ref.py).
struct FqP {
    FF_HD static constexpr uint32_t mod(int i) {
        constexpr uint32_t M[8] = {0xd87cfd47u, 0x3c208c16u, 0x6871ca8du, 0x97816a91u,
                                   0x8181585du, 0xb85045b6u, 0xe131a029u, 0x30644e72u};
        return M[i];
    }
    FF_HD static constexpr uint32_t inv() { return 0xe4866389u; }
    FF_HD static constexpr uint32_t r2(int i) {
        constexpr uint32_t M[8] = {0x538afa89u, 0xf32cfc5bu, 0xd44501fbu, 0xb5e71911u,
                                   0x0a417ff6u, 0x47ab1effu, 0xcab8351fu, 0x06d89f71u};
        return M[i];
    }
    FF_HD static constexpr uint32_t one(int i) {
        constexpr uint32_t M[8] = {0xc58f0d9du, 0xd35d438du, 0xf5c70b3du, 0x0a78eb28u,
                                   0x7879462cu, 0x666ea36fu, 0x9a07df2fu, 0x0e0a77c1u};
        return M[i];
    }
};
struct FrP {
    FF_HD static constexpr uint32_t mod(int i) {
        constexpr uint32_t M[8] = {0xf0000001u, 0x43e1f593u, 0x79b97091u, 0x2833e848u,
                                   0x8181585du, 0xb85045b6u, 0xe131a029u, 0x30644e72u};
        return M[i];
    }
    FF_HD static constexpr uint32_t inv() { return 0xefffffffu; }
    FF_HD static constexpr uint32_t r2(int i) {
        constexpr uint32_t M[8] = {0xae216da7u, 0x1bb8e645u, 0xe35c59e3u, 0x53fe3ab1u,
                                   0x53bb8085u, 0x8c49833du, 0x7f4e44a5u, 0x0216d0b1u};
        return M[i];
    }
    FF_HD static constexpr uint32_t one(int i) {
        constexpr uint32_t M[8] = {0x4ffffffbu, 0xac96341cu, 0x9f60cd29u, 0x36fc7695u,
                                   0x7879462eu, 0x666ea36fu, 0x9a07df2fu, 0x0e0a77c1u};
        return M[i];
    }
};

// ---- generic ops ------------------------------------------------------------
template <class C> FF_HD void ff_set_one(fp256& o) {
    for (int i = 0; i < 8; i++) o.l[i] = C::one(i);
}
FF_HD void ff_set_zero(fp256& o) {
    for (int i = 0; i < 8; i++) o.l[i] = 0;
}
FF_HD bool ff_is_zero(const fp256& a) {
    uint32_t x = 0;
    for (int i = 0; i < 8; i++) x |= a.l[i];
    return x == 0;
}
FF_HD bool ff_eq(const fp256& a, const fp256& b) {
    uint32_t x = 0;
    for (int i = 0; i < 8; i++) x |= a.l[i] ^ b.l[i];
    return x == 0;
}
template <class C> FF_HD bool ff_geq_mod(const fp256& a) {
    for (int i = 7; i >= 0; i--) {
        if (a.l[i] != C::mod(i)) return a.l[i] > C::mod(i);
    }
    return true;
}
// o = a - b, returns borrow
FF_HD uint32_t ff_sub_raw(fp256& o, const fp256& a, const fp256& b) {
    uint64_t brw = 0;
    for (int i = 0; i < 8; i++) {
        uint64_t d = (uint64_t)a.l[i] - b.l[i] - brw;
        o.l[i] = (uint32_t)d;
        brw = (d >> 32) & 1;
    }
    return (uint32_t)brw;
}
FF_HD uint32_t ff_add_raw(fp256& o, const fp256& a, const fp256& b) {
    uint64_t c = 0;
    for (int i = 0; i < 8; i++) {
        c += (uint64_t)a.l[i] + b.l[i];
        o.l[i] = (uint32_t)c;
        c >>= 32;
    }
    return (uint32_t)c;
}
template <class C> FF_HD void ff_cond_sub_mod(fp256& o, uint32_t extra) {
    if (extra || ff_geq_mod<C>(o)) {
        uint64_t brw = 0;
        for (int i = 0; i < 8; i++) {
            uint64_t d = (uint64_t)o.l[i] - C::mod(i) - brw;
            o.l[i] = (uint32_t)d;
            brw = (d >> 32) & 1;
        }
    }
}
template <class C> FF_HD void ff_add(fp256& o, const fp256& a, const fp256& b) {
    uint32_t c = ff_add_raw(o, a, b);
    ff_cond_sub_mod<C>(o, c);
}
template <class C> FF_HD void ff_sub(fp256& o, const fp256& a, const fp256& b) {
    if (ff_sub_raw(o, a, b)) {
        uint64_t c = 0;
        for (int i = 0; i < 8; i++) {
            c += (uint64_t)o.l[i] + C::mod(i);
            o.l[i] = (uint32_t)c;
            c >>= 32;
        }
    }
}
template <class C> FF_HD void ff_neg(fp256& o, const fp256& a) {
    if (ff_is_zero(a)) { o = a; return; }
    fp256 m;
    for (int i = 0; i < 8; i++) m.l[i] = C::mod(i);
    ff_sub_raw(o, m, a);
}
FF_HD void ff_dbl_raw(fp256& o, const fp256& a) { ff_add_raw(o, a, a); }

// CIOS Montgomery multiplication, 8x32 limbs, 64-bit accumulators.
// (Host path + reference; the device path below dispatches to the asm
// column form, which must stay bit-identical to this.)
template <class C> FF_HD void ff_mul_cios(fp256& o, const fp256& a, const fp256& b) {
    uint32_t t[10];
    for (int i = 0; i < 10; i++) t[i] = 0;
    for (int i = 0; i < 8; i++) {
        uint64_t cc = 0;
        const uint32_t ai = a.l[i];
        for (int j = 0; j < 8; j++) {
            uint64_t x = (uint64_t)ai * b.l[j] + t[j] + (uint32_t)cc;
            t[j] = (uint32_t)x;
            cc = x >> 32;
        }
        uint64_t x = (uint64_t)t[8] + (uint32_t)cc;
        t[8] = (uint32_t)x;
        t[9] = (uint32_t)(x >> 32);
        const uint32_t m = t[0] * C::inv();
        uint64_t x2 = (uint64_t)m * C::mod(0) + t[0];
        cc = x2 >> 32;
        for (int j = 1; j < 8; j++) {
            x2 = (uint64_t)m * C::mod(j) + t[j] + (uint32_t)cc;
            t[j - 1] = (uint32_t)x2;
            cc = x2 >> 32;
        }
        x2 = (uint64_t)t[8] + (uint32_t)cc;
        t[7] = (uint32_t)x2;
        t[8] = t[9] + (uint32_t)(x2 >> 32);
    }
    for (int i = 0; i < 8; i++) o.l[i] = t[i];
    ff_cond_sub_mod<C>(o, t[8]);
}
// ---- device multiply: hand-scheduled column Montgomery --------------------
// On gfx950 the asm column form runs at ~135 G Fq-mul/s vs ~103 for the
// compiled CIOS (measured, tools/microbench.hip) and is bit-identical
// (256K-lane chained check + the whole GPU parity suite). Define
// SPECTRE_NO_ASM_MUL to fall back to the C form for debugging/A-B.
#if defined(__HIP_DEVICE_COMPILE__) && !defined(SPECTRE_NO_ASM_MUL)
#define FF_USE_ASM_MUL 1
#endif

#if defined(FF_USE_ASM_MUL)
// Single product: acc += a*b (64-bit mad), ovf += carry-out. Explicit
// carry-mask pairs ("=s"): a vcc clobber makes hipcc pad every block
// boundary with s_nop hazards.
// MEASURED NEGATIVE RESULT (r2): fusing 2/4 products per asm block to cut
// those boundary s_nops produces WRONG results at full-range operands — a
// v_mad_u64_u32 reading the 64-bit acc written by a mad one instruction
// earlier needs >= 2 intervening issue slots (the compiler's conservative
// inter-block s_nop was supplying exactly that); and the s_nops are hidden
// by co-resident waves anyway (grouped rate unchanged at 135 G/s). Keep
// one mad+addc pair per block.
__device__ __forceinline__ void ff_mad64_(uint64_t& acc, uint32_t& ovf,
                                          uint32_t a, uint32_t b) {
    uint64_t c;
    asm volatile("v_mad_u64_u32 %0, %2, %3, %4, %0\n\t"
                 "v_addc_co_u32 %1, %2, 0, %1, %2"
                 : "+v"(acc), "+v"(ovf), "=s"(c)
                 : "v"(a), "v"(b));
}
__device__ __forceinline__ void ff_mad64_s_(uint64_t& acc, uint32_t& ovf,
                                            uint32_t a, uint32_t b_uniform) {
    uint64_t c;
    asm volatile("v_mad_u64_u32 %0, %2, %3, %4, %0\n\t"
                 "v_addc_co_u32 %1, %2, 0, %1, %2"
                 : "+v"(acc), "+v"(ovf), "=s"(c)
                 : "v"(a), "s"(b_uniform));
}
template <class C, int K>
__device__ __forceinline__ void detail_ff_low_cols(uint64_t& acc,
                                                   uint32_t& ovf,
                                                   const uint32_t* a,
                                                   const uint32_t* b,
                                                   uint32_t* m) {
    if constexpr (K < 8) {
#pragma unroll
        for (int i = 0; i <= K; i++) ff_mad64_(acc, ovf, a[i], b[K - i]);
#pragma unroll
        for (int i = 0; i < K; i++) ff_mad64_s_(acc, ovf, m[i], C::mod(K - i));
        m[K] = (uint32_t)acc * C::inv();
        ff_mad64_s_(acc, ovf, m[K], C::mod(0));
        acc = (acc >> 32) | ((uint64_t)ovf << 32);
        ovf = 0;
        detail_ff_low_cols<C, K + 1>(acc, ovf, a, b, m);
    }
}
template <class C, int K>
__device__ __forceinline__ void detail_ff_high_cols(uint64_t& acc,
                                                    uint32_t& ovf,
                                                    const uint32_t* a,
                                                    const uint32_t* b,
                                                    const uint32_t* m,
                                                    uint32_t* out) {
    if constexpr (K < 15) {
#pragma unroll
        for (int i = K - 7; i < 8; i++) ff_mad64_(acc, ovf, a[i], b[K - i]);
#pragma unroll
        for (int i = K - 7; i < 8; i++) ff_mad64_s_(acc, ovf, m[i], C::mod(K - i));
        out[K - 8] = (uint32_t)acc;
        acc = (acc >> 32) | ((uint64_t)ovf << 32);
        ovf = 0;
        detail_ff_high_cols<C, K + 1>(acc, ovf, a, b, m, out);
    }
}
// Product-scanning Montgomery: low columns annihilated with m_k as they
// complete, high columns emit the result; ovf counts 64-bit carry-outs per
// column (<= 16 products). Result < 2p, one conditional subtract.
template <class C>
__device__ __forceinline__ void ff_mul(fp256& o, const fp256& A,
                                       const fp256& B) {
    uint32_t m[8];
    uint64_t acc = 0;
    uint32_t ovf = 0;
    detail_ff_low_cols<C, 0>(acc, ovf, A.l, B.l, m);
    detail_ff_high_cols<C, 8>(acc, ovf, A.l, B.l, m, o.l);
    o.l[7] = (uint32_t)acc;
    // bits >= 256 provably zero (result < 2p < 2^255); high half passed to
    // the conditional subtract so a violated precondition reduces loudly
    // rather than truncating silently.
    ff_cond_sub_mod<C>(o, (uint32_t)(acc >> 32));
}
#else
template <class C> FF_HD void ff_mul(fp256& o, const fp256& a, const fp256& b) {
    ff_mul_cios<C>(o, a, b);
}
#endif

template <class C> FF_HD void ff_sqr(fp256& o, const fp256& a) { ff_mul<C>(o, a, a); }

// Montgomery conversion
template <class C> FF_HD void ff_to_mont(fp256& o, const fp256& a_canon) {
    fp256 r2;
    for (int i = 0; i < 8; i++) r2.l[i] = C::r2(i);
    ff_mul<C>(o, a_canon, r2);
}
template <class C> FF_HD void ff_from_mont(fp256& o, const fp256& a) {
    fp256 one;
    ff_set_zero(one);
    one.l[0] = 1;
    ff_mul<C>(o, a, one);
}

// a^e, e a canonical 256-bit little-endian exponent; a, out Montgomery.
template <class C> FF_HD void ff_pow(fp256& o, const fp256& a, const fp256& e) {
    fp256 acc, base = a;
    ff_set_one<C>(acc);
    for (int i = 0; i < 256; i++) {
        if ((e.l[i >> 5] >> (i & 31)) & 1) ff_mul<C>(acc, acc, base);
        ff_sqr<C>(base, base);
    }
    o = acc;
}
// a^e for small 32-bit exponent (twiddle powers)
template <class C> FF_HD void ff_pow_u32(fp256& o, const fp256& a, uint32_t e) {
    fp256 acc, base = a;
    ff_set_one<C>(acc);
    while (e) {
        if (e & 1) ff_mul<C>(acc, acc, base);
        e >>= 1;
        if (e) ff_sqr<C>(base, base);
    }
    o = acc;
}
// Fermat inverse: a^(m-2)
template <class C> FF_HD void ff_inv(fp256& o, const fp256& a) {
    fp256 e;
    for (int i = 0; i < 8; i++) e.l[i] = C::mod(i);
    // subtract 2 (low limb of both moduli is odd and >= 2^0+...; handle borrow anyway)
    uint64_t d = (uint64_t)e.l[0] - 2;
    e.l[0] = (uint32_t)d;
    uint64_t brw = (d >> 32) & 1;
    for (int i = 1; i < 8 && brw; i++) {
        d = (uint64_t)e.l[i] - brw;
        e.l[i] = (uint32_t)d;
        brw = (d >> 32) & 1;
    }
    ff_pow<C>(o, a, e);
}

FF_HD void ff_from_bytes(fp256& o, const uint8_t* b) { memcpy(o.l, b, 32); }
FF_HD void ff_to_bytes(uint8_t* b, const fp256& a) { memcpy(b, a.l, 32); }

using Fq = FqP;
using Fr = FrP;
