// ntt.hip — radix-2 NTT over BN254 Fr for gfx950 (MI355X).
//
// Computes what halo2's best_fft / EvaluationDomain compute (the plain DFT
// out[j] = Sum_i a[i] omega^(ij); inverse = DFT with omega^{-1} then * n^{-1};
// coset = multiply by coset_gen^i before (forward) / after (inverse)).
//
// MI355X-native structure (four-step / Bailey decomposition, n = n1*n2,
// n2 = min(2^12, n), so each sub-transform fits LDS and the whole transform
// is TWO global passes instead of log2(n) — the difference between ~1 GiB
// and ~11.5 GiB of HBM traffic at n=2^23):
//   pass A (k_ntt_col, only when n1 > 1): one workgroup per column i2:
//     column DFT over i1 (stride n2), in-LDS DIF (bit-reversed output order
//     is folded into the store index, costing nothing extra — the column
//     store is strided anyway), fused inter-pass twiddle omega^(i2*t1) and
//     optional forward-coset pre-multiply.
//   pass B (k_ntt_row): one workgroup per row t1: contiguous (coalesced)
//     row DFT over i2, in-LDS DIF, store transposed to out[t2*n1 + t1] with
//     fused n^{-1} scaling and optional inverse-coset post-multiply.
// Twiddle/coset powers g^e (e < n <= 2^24) resolve via two 4096-entry tables:
//   g^e = T1[e & 0xfff] * T2[e >> 12]   (one mul + two cached loads)
// built once per (omega, log_n) plan and cached on device.
//
// For log_n in [25, 28] (the aggregation circuits' extended domains reach
// 2^25..2^26; Fr's 2-adicity caps at 2^28) a THREE-pass variant (k_ntt_axis)
// decomposes n = A*B*C with each axis <= 2^12:
//   X[tA + A*tB + A*B*tC] = DFT_C(twiddle(DFT_B(twiddle(DFT_A(x)))))
// with post-twiddles w^(tA*(b*C+c)) after pass 1 and w^(A*tB*c) after pass 2
// (derivation in the k_ntt_axis comment).
//
// All Fr math is 8x32-limb Montgomery (ff.hpp); data stays in Montgomery form
// end-to-end exactly as halo2 holds its &[Fr] slices.
//
// This TU keeps the C CIOS multiply: the asm column multiply regressed the
// 1024-thread LDS kernels ~7% (2^23 fwd 2.59 -> 2.77 ms — register pressure
// at the 4-waves/SIMD occupancy these kernels need), while it wins ~28% in
// the 256-thread MSM kernels. Per-TU choice, measured r2.
#ifndef SPECTRE_NTT_ASMMUL  // variant: -DSPECTRE_NTT_ASMMUL re-tests the
#define SPECTRE_NO_ASM_MUL 1  // asm multiply under new block geometries
#endif
#include "internal.hpp"

#define THREADS 256       // pow-table builder
#define NTT_THREADS 1024  // NTT passes: 16 waves/block so one LDS-resident
                          // block still puts 4 waves on every SIMD
#ifdef SPECTRE_NTT_RADIX2
static constexpr bool kForceRadix2 = true;   // A/B variant build
#else
static constexpr bool kForceRadix2 = false;
#endif
#define TW_LOW_BITS 12
#define TW_LOW_MASK 0xfffu

__device__ __forceinline__ uint32_t bitrev(uint32_t x, uint32_t bits) {
    return bits ? (__brev(x) >> (32 - bits)) : 0;
}
// LDS layout: a 32-B element spans 8 banks, so a 16-lane ds_read_b128 group
// covers only 8 bank-octets — inherent 2-8x conflict at power-of-two
// butterfly strides (measured SQ_LDS_BANK_CONFLICT/IDX_ACTIVE = 0.85 with a
// single fp256 array). Instead each element is stored as TWO 16-B halves in
// separate regions (lo at [i], hi at [H+i]): consecutive elements stride 4
// banks and one lane group covers all 64 banks.
#define LDS_BYTES(L) ((uint32_t)(L) * 32u)
__device__ __forceinline__ void lds_ld(const uint4* lds4, uint32_t H,
                                       uint32_t i, fp256& o) {
    uint4 lo = lds4[i], hi = lds4[H + i];
    memcpy(&o.l[0], &lo, 16);
    memcpy(&o.l[4], &hi, 16);
}
__device__ __forceinline__ void lds_st(uint4* lds4, uint32_t H, uint32_t i,
                                       const fp256& v) {
    uint4 lo, hi;
    memcpy(&lo, &v.l[0], 16);
    memcpy(&hi, &v.l[4], 16);
    lds4[i] = lo;
    lds4[H + i] = hi;
}
// g^e via the 2D power table (e < n <= 2^24)
__device__ __forceinline__ void tw_lookup(fp256& o, const fp256* T1,
                                          const fp256* T2, uint32_t e) {
    ff_mul<Fr>(o, T1[e & TW_LOW_MASK], T2[e >> TW_LOW_BITS]);
}

// out[j] = base^j, j < count
__global__ void k_pow_table(fp256 base, fp256* __restrict__ out,
                            uint32_t count) {
    uint32_t j = blockIdx.x * blockDim.x + threadIdx.x;
    if (j >= count) return;
    fp256 v;
    ff_pow_u32<Fr>(v, base, j);
    out[j] = v;
}

// in-LDS DIF butterflies over L = 2^logL elements; twL[j] = (root)^j, j < L/2.
// On exit lds[s] holds DFT output index bitrev(s, logL).
//
// Levels run in FUSED PAIRS (radix-4 rounds): a thread loads the quad
// (i0, i0+h/2, i0+h, i0+3h/2), applies level h then level h/2 in registers,
// and writes back once — HALF the LDS round trips and HALF the barriers of
// per-level radix-2, with the exact same field operations (bit-identical
// results; finite-field ops are exact). An odd level count runs one plain
// radix-2 level first. Build with -DSPECTRE_NTT_RADIX2 to restore the
// per-level schedule (A/B variant).
// R4 is a compile-time variant: the radix-4 quad path costs VGPRs/code even
// when branched over at run time (measured 2^20 0.40 -> 0.46 ms with a
// run-time flag), so the host launches the <true> instantiation only when
// the tile gives every thread a quad (L >= 4*threads).
template <bool R4>
__device__ void lds_dif(uint4* lds4, uint32_t H, const fp256* __restrict__ twL,
                        uint32_t logL) {
    const uint32_t L = 1u << logL;
    uint32_t h = L >> 1;
    if (!R4 || (logL & 1))  // radix-2: all levels, or one to make pairs even
    {
        for (; h >= 1; h >>= 1) {
            const uint32_t stride = (L >> 1) / h;
            for (uint32_t p = threadIdx.x; p < (L >> 1); p += blockDim.x) {
                uint32_t blk = p / h, j = p % h;
                uint32_t i0 = blk * 2 * h + j, i1 = i0 + h;
                fp256 u, v, t, w;
                lds_ld(lds4, H, i0, u);
                lds_ld(lds4, H, i1, v);
                ff_add<Fr>(w, u, v);
                lds_st(lds4, H, i0, w);
                ff_sub<Fr>(t, u, v);
                ff_mul<Fr>(t, t, twL[(uint64_t)j * stride]);
                lds_st(lds4, H, i1, t);
            }
            __syncthreads();
            if constexpr (R4) { h >>= 1; break; }
        }
    }
    if constexpr (!R4) return;
    for (; h >= 2; h >>= 2) {
        const uint32_t hh = h >> 1;               // second level of the pair
        const uint32_t s_h = (L >> 1) / h;        // level-h twiddle stride
        for (uint32_t p = threadIdx.x; p < (L >> 2); p += blockDim.x) {
            const uint32_t blk = p / hh, j = p % hh;
            const uint32_t i0 = blk * 2 * h + j;
            fp256 a, b, c, d, t;
            lds_ld(lds4, H, i0, a);
            lds_ld(lds4, H, i0 + hh, b);
            lds_ld(lds4, H, i0 + h, c);
            lds_ld(lds4, H, i0 + h + hh, d);
            // level h: pairs (a,c) at offset j and (b,d) at offset j+hh
            fp256 a1, b1, c1, d1;
            ff_add<Fr>(a1, a, c);
            ff_sub<Fr>(t, a, c);
            ff_mul<Fr>(c1, t, twL[(uint64_t)j * s_h]);
            ff_add<Fr>(b1, b, d);
            ff_sub<Fr>(t, b, d);
            ff_mul<Fr>(d1, t, twL[(uint64_t)(j + hh) * s_h]);
            // level h/2: pairs (a1,b1) and (c1,d1), both at offset j,
            // twiddle stride 2*s_h
            ff_add<Fr>(a, a1, b1);
            ff_sub<Fr>(t, a1, b1);
            ff_mul<Fr>(b, t, twL[(uint64_t)j * 2 * s_h]);
            ff_add<Fr>(c, c1, d1);
            ff_sub<Fr>(t, c1, d1);
            ff_mul<Fr>(d, t, twL[(uint64_t)j * 2 * s_h]);
            lds_st(lds4, H, i0, a);
            lds_st(lds4, H, i0 + hh, b);
            lds_st(lds4, H, i0 + h, c);
            lds_st(lds4, H, i0 + h + hh, d);
        }
        __syncthreads();
    }
}

// pass A: column DFTs. grid.x = n2; LDS = n1 elements.
template <bool R4>
__global__ __launch_bounds__(NTT_THREADS) void k_ntt_col(
                          const fp256* __restrict__ in, fp256* __restrict__ out,
                          const fp256* __restrict__ tw1,
                          const fp256* __restrict__ T1,
                          const fp256* __restrict__ T2,
                          const fp256* __restrict__ cT1,
                          const fp256* __restrict__ cT2, uint32_t log_n1,
                          uint32_t log_n2) {
    extern __shared__ uint4 lds4[];
    const uint32_t n1 = 1u << log_n1;
    const uint32_t n2 = 1u << log_n2;
    const uint32_t H = n1;
    const uint32_t c = blockIdx.x;
    for (uint32_t s = threadIdx.x; s < n1; s += blockDim.x) {
        fp256 v = in[(uint64_t)s * n2 + c];
        if (cT1) {  // forward coset: multiply by g^(global index)
            fp256 f;
            tw_lookup(f, cT1, cT2, s * n2 + c);
            ff_mul<Fr>(v, v, f);
        }
        lds_st(lds4, H, s, v);
    }
    __syncthreads();
    lds_dif<R4>(lds4, H, tw1, log_n1);
    // store with inter-pass twiddle omega^(c * t1), t1 = bitrev(s)
    for (uint32_t s = threadIdx.x; s < n1; s += blockDim.x) {
        uint32_t t1 = bitrev(s, log_n1);
        fp256 f, v;
        tw_lookup(f, T1, T2, c * t1);  // c*t1 < n2*n1 = n
        fp256 x;
        lds_ld(lds4, H, s, x);
        ff_mul<Fr>(v, x, f);
        out[(uint64_t)t1 * n2 + c] = v;
    }
}

// pass B: row DFTs + transposed store. grid.x = n1; LDS = n2 elements.
// in == out is safe only when n1 == 1 (single workgroup).
template <bool R4>
__global__ __launch_bounds__(NTT_THREADS) void k_ntt_row(
                          const fp256* __restrict__ in, fp256* __restrict__ out,
                          const fp256* __restrict__ tw2,
                          const fp256* __restrict__ cT1,
                          const fp256* __restrict__ cT2, int coset_on_load,
                          fp256 scale, int apply_scale, uint32_t log_n1,
                          uint32_t log_n2) {
    extern __shared__ uint4 lds4[];
    const uint32_t n1 = 1u << log_n1;
    const uint32_t n2 = 1u << log_n2;
    const uint32_t H = n2;
    const uint32_t r = blockIdx.x;
    for (uint32_t s = threadIdx.x; s < n2; s += blockDim.x) {
        fp256 v = in[(uint64_t)r * n2 + s];
        if (cT1 && coset_on_load) {  // 1-pass forward coset (r == 0)
            fp256 f;
            tw_lookup(f, cT1, cT2, s);
            ff_mul<Fr>(v, v, f);
        }
        lds_st(lds4, H, s, v);
    }
    __syncthreads();
    lds_dif<R4>(lds4, H, tw2, log_n2);
    for (uint32_t s = threadIdx.x; s < n2; s += blockDim.x) {
        uint32_t t2 = bitrev(s, log_n2);
        fp256 v;
        lds_ld(lds4, H, s, v);
        if (apply_scale) ff_mul<Fr>(v, v, scale);
        if (cT1 && !coset_on_load) {  // inverse coset: g^(output index)
            fp256 f;
            tw_lookup(f, cT1, cT2, t2 * n1 + r);
            ff_mul<Fr>(v, v, f);
        }
        out[(uint64_t)t2 * n1 + r] = v;
    }
}

// generic strided-axis DFT for the three-pass (log_n > 24) path.
// Transform m (grid.x) covers element j at
//   pos(m, j) = (m >> logS) << (logL + logS) | (m & (S-1)) | j*S-positioned,
// i.e. base + j*S with base = (m >> logS)*L*S + (m & (S-1)).
// Derivation (i = a*BC + b*C + c, t = tA + A*tB + AB*tC):
//   i*t mod n = a*BC*tA + b*C*(tA + A*tB) + c*(tA + A*tB + AB*tC)
// so pass1 (axis a, root w^(BC)) is followed by twiddle w^(tA*(b*C+c)) =
// w^(t*m); pass2 (axis b, root w^(AC)) by w^(A*tB*c) = w^((t<<logA)*(m&(C-1)));
// pass3 (axis c, root w^(AB)) scatters to out[tA + A*tB + AB*t] and carries
// the n^{-1} scale / inverse-coset factor. All exponents < n <= 2^28 (u32).
template <bool R4>
__global__ __launch_bounds__(NTT_THREADS) void k_ntt_axis(
    const fp256* __restrict__ in, fp256* __restrict__ out,
    const fp256* __restrict__ twL, const fp256* __restrict__ T1,
    const fp256* __restrict__ T2, const fp256* __restrict__ cT1,
    const fp256* __restrict__ cT2, int coset_on_load, fp256 scale,
    int apply_scale, uint32_t logL, uint32_t logS, uint32_t post_mode,
    uint32_t post_logA, uint32_t post_logC, int final_scatter, uint32_t fin_A,
    uint32_t fin_B) {
    extern __shared__ uint4 lds4[];
    const uint32_t L = 1u << logL;
    const uint32_t S = 1u << logS;
    const uint32_t H = L;
    const uint32_t m = blockIdx.x;
    const uint64_t base =
        ((uint64_t)(m >> logS) << (logL + logS)) + (m & (S - 1));
    for (uint32_t s = threadIdx.x; s < L; s += blockDim.x) {
        const uint64_t pos = base + (uint64_t)s * S;
        fp256 v = in[pos];
        if (cT1 && coset_on_load) {  // forward coset: g^(original index)
            fp256 f;
            tw_lookup(f, cT1, cT2, (uint32_t)pos);
            ff_mul<Fr>(v, v, f);
        }
        lds_st(lds4, H, s, v);
    }
    __syncthreads();
    lds_dif<R4>(lds4, H, twL, logL);
    for (uint32_t s = threadIdx.x; s < L; s += blockDim.x) {
        const uint32_t t = bitrev(s, logL);
        fp256 v, f;
        lds_ld(lds4, H, s, v);
        if (post_mode == 1) {
            tw_lookup(f, T1, T2, t * m);
            ff_mul<Fr>(v, v, f);
        } else if (post_mode == 2) {
            tw_lookup(f, T1, T2, (t << post_logA) * (m & ((1u << post_logC) - 1)));
            ff_mul<Fr>(v, v, f);
        }
        if (apply_scale) ff_mul<Fr>(v, v, scale);
        uint64_t opos;
        if (final_scatter) {
            const uint32_t tA = m / fin_B, tB = m % fin_B;
            opos = (uint64_t)tA + (uint64_t)fin_A * tB +
                   (uint64_t)fin_A * fin_B * t;
            if (cT1 && !coset_on_load) {  // inverse coset: g^(output index)
                tw_lookup(f, cT1, cT2, (uint32_t)opos);
                ff_mul<Fr>(v, v, f);
            }
        } else {
            opos = base + (uint64_t)t * S;
        }
        out[opos] = v;
    }
}

// ---- gate-expression evaluator (quotient phase) -------------
// One launch evaluates a whole custom-gate expression over every row: a
// stack machine whose top-of-stack lives in registers and whose lower slots
// live in LDS (a register-indexed array would spill to scratch — LDS slots
// are strided [slot][tid] so access is conflict-free). The program is tiny
// and wave-uniform; columns are read with rotation (row + rot*rot_scale)
// mod n (n is a power of two).
#define GATE_THREADS 256
__global__ __launch_bounds__(GATE_THREADS) void k_fr_gate_eval(
    const fp256* const* __restrict__ cols, const fp256* __restrict__ consts,
    const uint32_t* __restrict__ prog, uint32_t nops, uint64_t n,
    uint32_t rot_scale, int use_y, fp256 y, fp256* __restrict__ out) {
    extern __shared__ uint4 lds4[];
    const uint64_t row = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (row >= n) return;
    const uint32_t T = blockDim.x;
    const uint64_t mask = n - 1;
    fp256 tos;
    ff_set_zero(tos);
    int depth = 0;
    auto lds_slot_st = [&](int s, const fp256& v) {
        uint4 lo, hi;
        memcpy(&lo, &v.l[0], 16);
        memcpy(&hi, &v.l[4], 16);
        lds4[(2 * s) * T + threadIdx.x] = lo;
        lds4[(2 * s + 1) * T + threadIdx.x] = hi;
    };
    auto lds_slot_ld = [&](int s, fp256& v) {
        uint4 lo = lds4[(2 * s) * T + threadIdx.x];
        uint4 hi = lds4[(2 * s + 1) * T + threadIdx.x];
        memcpy(&v.l[0], &lo, 16);
        memcpy(&v.l[4], &hi, 16);
    };
    for (uint32_t pc = 0; pc < nops; pc++) {
        const uint32_t op = prog[3 * pc];
        const uint32_t a = prog[3 * pc + 1];
        const int32_t b = (int32_t)prog[3 * pc + 2];
        if (op == SPECTRE_GATE_OP_COL || op == SPECTRE_GATE_OP_CONST) {
            if (depth >= 1) lds_slot_st(depth - 1, tos);
            if (op == SPECTRE_GATE_OP_COL) {
                const uint64_t idx =
                    (row + (uint64_t)((int64_t)b * rot_scale + (int64_t)n)) &
                    mask;
                tos = cols[a][idx];
            } else {
                tos = consts[a];
            }
            depth++;
        } else if (op == SPECTRE_GATE_OP_NEG) {
            ff_neg<Fr>(tos, tos);
        } else {  // binary: (second) op (top)
            fp256 lhs;
            lds_slot_ld(depth - 2, lhs);
            if (op == SPECTRE_GATE_OP_ADD) ff_add<Fr>(tos, lhs, tos);
            else if (op == SPECTRE_GATE_OP_SUB) ff_sub<Fr>(tos, lhs, tos);
            else ff_mul<Fr>(tos, lhs, tos);
            depth--;
        }
    }
    if (use_y) {
        fp256 acc = out[row];
        ff_mul<Fr>(acc, acc, y);
        ff_add<Fr>(tos, acc, tos);
    }
    out[row] = tos;
}

int fr_gate_eval_device(spectre_gpu_ctx* ctx, int dev,
                        const fp256* const* cols, uint32_t ncols,
                        const fp256* consts, uint32_t nconst,
                        const uint32_t* program, uint32_t nops, uint64_t n,
                        uint32_t rot_scale, const fp256* y, fp256* d_out) {
    DeviceState& ds = ctx->devs[dev];
    HIP_TRY(hipSetDevice(ds.device_id));
    if (n == 0 || (n & (n - 1)) != 0) {
        set_err("gate_eval: n must be a nonzero power of two");
        return -3;
    }
    // validate the program and compute its maximum stack depth
    int depth = 0, maxd = 0;
    for (uint32_t pc = 0; pc < nops; pc++) {
        const uint32_t op = program[3 * pc];
        const uint32_t a = program[3 * pc + 1];
        const int32_t b = (int32_t)program[3 * pc + 2];
        switch (op) {
            case SPECTRE_GATE_OP_COL:
                if (a >= ncols) { set_err("gate_eval: col %u >= %u", a, ncols); return -3; }
                if ((uint64_t)((int64_t)b * rot_scale < 0
                                   ? -(int64_t)b * rot_scale
                                   : (int64_t)b * rot_scale) >= n) {
                    set_err("gate_eval: rotation %d * %u out of range", b, rot_scale);
                    return -3;
                }
                depth++;
                break;
            case SPECTRE_GATE_OP_CONST:
                if (a >= nconst) { set_err("gate_eval: const %u >= %u", a, nconst); return -3; }
                depth++;
                break;
            case SPECTRE_GATE_OP_NEG:
                if (depth < 1) { set_err("gate_eval: NEG on empty stack"); return -3; }
                break;
            case SPECTRE_GATE_OP_ADD:
            case SPECTRE_GATE_OP_SUB:
            case SPECTRE_GATE_OP_MUL:
                if (depth < 2) { set_err("gate_eval: binary op underflow at pc %u", pc); return -3; }
                depth--;
                break;
            default:
                set_err("gate_eval: bad opcode %u at pc %u", op, pc);
                return -3;
        }
        if (depth > maxd) maxd = depth;
        if (maxd > SPECTRE_GATE_MAX_DEPTH) {
            set_err("gate_eval: stack depth %d > %d", maxd, SPECTRE_GATE_MAX_DEPTH);
            return -3;
        }
    }
    if (depth != 1) {
        set_err("gate_eval: program leaves %d values on the stack (need 1)", depth);
        return -3;
    }
    // device staging: [col ptrs][constants][program], one cached buffer
    const size_t ptr_b = (size_t)ncols * sizeof(fp256*);
    const size_t con_b = (size_t)nconst * sizeof(fp256);
    const size_t prg_b = (size_t)nops * 12;
    const size_t need = ptr_b + con_b + prg_b;
    if (ds.gate_cap < need) {
        if (ds.d_gate) { (void)hipFree(ds.d_gate); ds.d_gate = nullptr; }
        ds.gate_cap = 0;
        HIP_TRY(hipMalloc(&ds.d_gate, need));
        ds.gate_cap = need;
    }
    HIP_TRY(hipMemcpyAsync(ds.d_gate, cols, ptr_b, hipMemcpyHostToDevice,
                           ds.stream));
    if (con_b)
        HIP_TRY(hipMemcpyAsync(ds.d_gate + ptr_b, consts, con_b,
                               hipMemcpyHostToDevice, ds.stream));
    HIP_TRY(hipMemcpyAsync(ds.d_gate + ptr_b + con_b, program, prg_b,
                           hipMemcpyHostToDevice, ds.stream));
    fp256 yv;
    ff_set_zero(yv);
    if (y) yv = *y;
    const uint32_t lds_bytes =
        (maxd > 1 ? (uint32_t)(maxd - 1) * GATE_THREADS * 32u : 0u);
    hipLaunchKernelGGL(k_fr_gate_eval,
                       dim3((uint32_t)((n + GATE_THREADS - 1) / GATE_THREADS)),
                       dim3(GATE_THREADS), lds_bytes, ds.stream,
                       (const fp256* const*)ds.d_gate,
                       (const fp256*)(ds.d_gate + ptr_b),
                       (const uint32_t*)(ds.d_gate + ptr_b + con_b), nops, n,
                       rot_scale, y ? 1 : 0, yv, d_out);
    HIP_TRY(hipStreamSynchronize(ds.stream));
    HIP_TRY(hipGetLastError());
    return 0;
}

// ---- pointwise Fr vector ops (quotient-phase gate-eval glue) --------------
__global__ void k_fr_vec_op(int op, const fp256* __restrict__ a,
                            const fp256* __restrict__ b, fp256 c,
                            fp256* __restrict__ out, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    fp256 r, va = a[i];
    switch (op) {
        case SPECTRE_VEC_ADD: ff_add<Fr>(r, va, b[i]); break;
        case SPECTRE_VEC_SUB: ff_sub<Fr>(r, va, b[i]); break;
        case SPECTRE_VEC_MUL: ff_mul<Fr>(r, va, b[i]); break;
        case SPECTRE_VEC_SCALE: ff_mul<Fr>(r, va, c); break;
        default: {  // ADD_SCALED
            fp256 t;
            ff_mul<Fr>(t, b[i], c);
            ff_add<Fr>(r, va, t);
        }
    }
    out[i] = r;
}

int fr_vec_op_device(spectre_gpu_ctx* ctx, int dev, int op, const fp256* d_a,
                     const fp256* d_b, const fp256* c, fp256* d_out,
                     uint64_t n) {
    DeviceState& ds = ctx->devs[dev];
    HIP_TRY(hipSetDevice(ds.device_id));
    fp256 cv;
    ff_set_zero(cv);
    if (c) cv = *c;
    hipLaunchKernelGGL(k_fr_vec_op,
                       dim3((uint32_t)((n + THREADS - 1) / THREADS)),
                       dim3(THREADS), 0, ds.stream, op, d_a, d_b, cv, d_out,
                       n);
    HIP_TRY(hipStreamSynchronize(ds.stream));
    HIP_TRY(hipGetLastError());
    return 0;
}

// ---------------------------------------------------------------- host side
static void host_pow_u32(fp256& o, const fp256& a, uint32_t e) {
    ff_pow_u32<Fr>(o, a, e);
}

static int build_tables(DeviceState& ds, const fp256& base, fp256* out,
                        uint32_t count) {
    hipLaunchKernelGGL(k_pow_table, dim3((count + THREADS - 1) / THREADS),
                       dim3(THREADS), 0, ds.stream, base, out, count);
    return 0;
}

// build-or-get the cached twiddle plan for (omega, log_n)
static int get_plan(DeviceState& ds, const fp256& omega, uint32_t log_n,
                    NttPlan** out) {
    std::array<uint8_t, 40> key{};
    memcpy(key.data(), omega.l, 32);
    memcpy(key.data() + 32, &log_n, 4);
    auto it = ds.plans.find(key);
    if (it != ds.plans.end()) {
        *out = &it->second;
        return 0;
    }
    NttPlan p;
    const bool force3 = getenv("SPECTRE_NTT_FORCE3") != nullptr;
    const uint32_t t1n = (log_n < TW_LOW_BITS) ? (1u << log_n) : (1u << TW_LOW_BITS);
    const uint32_t t2n = (log_n > TW_LOW_BITS) ? (1u << (log_n - TW_LOW_BITS)) : 1;
    HIP_TRY(hipMalloc(&p.twB, ((uint64_t)t1n + t2n) * sizeof(fp256)));  // T1||T2
    fp256 wT2;
    host_pow_u32(wT2, omega, 1u << TW_LOW_BITS);
    build_tables(ds, omega, p.twB, t1n);
    build_tables(ds, wT2, p.twB + t1n, t2n);
    if (log_n > 24 || (force3 && log_n >= 3)) {
        // three-pass split n = A*B*C, each axis <= 2^12 (balanced), or
        // SPECTRE_NTT_SPLIT3=max: kA = 12 so pass A gets 4096-elem tiles
        // (radix-4 rounds) and B/C split the rest.
        // measured: kA=12 (radix-4 axis-A tiles) is -6% at 2^26, neutral at
        // 2^25 -> default for log_n >= 26; SPECTRE_NTT_SPLIT3=max/bal force.
        const char* s3 = getenv("SPECTRE_NTT_SPLIT3");
        const bool max3 = s3 ? (s3[0] == 'm') : (log_n >= 26);
        if (max3 && log_n > 14) {
            p.k1 = 12;
            const uint32_t rest = log_n - 12;
            p.k2 = (rest + 1) / 2;
            p.k3 = rest - p.k2;
        } else {
            const uint32_t q = log_n / 3, r = log_n % 3;
            p.k1 = q + (r > 0);  // kA
            p.k2 = q + (r > 1);  // kB
            p.k3 = q;            // kC
        }
        fp256 w1, w2, w3;
        host_pow_u32(w1, omega, 1u << (p.k2 + p.k3));  // w^(B*C)
        host_pow_u32(w2, omega, 1u << (p.k1 + p.k3));  // w^(A*C)
        host_pow_u32(w3, omega, 1u << (p.k1 + p.k2));  // w^(A*B)
        HIP_TRY(hipMalloc(&p.tw1, (1u << (p.k1 - 1)) * sizeof(fp256)));
        HIP_TRY(hipMalloc(&p.tw2, (1u << (p.k2 - 1)) * sizeof(fp256)));
        HIP_TRY(hipMalloc(&p.tw3, (1u << (p.k3 - 1)) * sizeof(fp256)));
        build_tables(ds, w1, p.tw1, 1u << (p.k1 - 1));
        build_tables(ds, w2, p.tw2, 1u << (p.k2 - 1));
        build_tables(ds, w3, p.tw3, 1u << (p.k3 - 1));
    } else {
        // 2-pass split, k2 >= k1 (the contiguous row pass gets the bigger
        // tile). Measured on MI355X: k2 = 12 (4096-elem row tiles -> radix-4
        // rounds) wins for log_n <= 20 (2^20 fwd 0.40 -> 0.32 ms) but LOSES
        // at 2^22 ((10,12) 1.40 vs balanced (11,11) 1.31 — the column pass's
        // longer stride outweighs the row gain), so: k2=12 up to log_n 20,
        // balanced above (log_n 23/24 balance to k2=12 anyway).
        // SPECTRE_NTT_SPLIT=max / =bal force either choice for A/B.
        const char* sp = getenv("SPECTRE_NTT_SPLIT");
        const bool maxk2 = sp ? (sp[0] == 'm') : (log_n <= 20);
        if (maxk2 && log_n > 12)
            p.k2 = 12;
        else
            p.k2 = log_n <= 12 ? log_n : (log_n + 1) / 2;
        p.k1 = log_n - p.k2;
        const uint32_t n1 = 1u << p.k1, n2 = 1u << p.k2;
        fp256 w1, w2;
        host_pow_u32(w1, omega, n2);  // root of column DFT
        host_pow_u32(w2, omega, n1);  // root of row DFT
        if (p.k1) {
            HIP_TRY(hipMalloc(&p.tw1, (n1 / 2) * sizeof(fp256)));
            build_tables(ds, w1, p.tw1, n1 / 2);
        }
        HIP_TRY(hipMalloc(&p.tw2, (n2 / 2 ? n2 / 2 : 1) * sizeof(fp256)));
        build_tables(ds, w2, p.tw2, n2 / 2 ? n2 / 2 : 1);
    }
    HIP_TRY(hipStreamSynchronize(ds.stream));
    HIP_TRY(hipGetLastError());
    auto res = ds.plans.emplace(key, p);
    *out = &res.first->second;
    return 0;
}

int ntt_device(spectre_gpu_ctx* ctx, int dev, fp256* d_data, uint32_t log_n,
               const fp256& omega, int inverse, const fp256* coset_gen) {
    DeviceState& ds = ctx->devs[dev];
    HIP_TRY(hipSetDevice(ds.device_id));
    if (log_n > 28) {  // BN254 Fr 2-adicity
        set_err("ntt: log_n %u > 28 unsupported (Fr 2-adicity)", log_n);
        return -3;
    }
    if (log_n == 0) return 0;  // DFT of size 1 is the identity; n^{-1} = 1, g^0 = 1
    const uint64_t n = 1ull << log_n;
    NttPlan* plan = nullptr;
    int rc = get_plan(ds, omega, log_n, &plan);
    if (rc) return rc;
    const uint32_t n1 = 1u << plan->k1, n2 = 1u << plan->k2;
    const uint32_t t1n = (log_n < TW_LOW_BITS) ? (uint32_t)n : (1u << TW_LOW_BITS);
    const uint32_t t2n = (log_n > TW_LOW_BITS) ? (uint32_t)(n >> TW_LOW_BITS) : 1;
    // scratch
    if (ds.ntt_cap < n) {
        if (ds.d_ntt_tmp) (void)hipFree(ds.d_ntt_tmp);
        HIP_TRY(hipMalloc(&ds.d_ntt_tmp, n * sizeof(fp256)));
        ds.ntt_cap = n;
    }
    // coset power tables (per call; tiny)
    fp256* cT1 = nullptr;
    fp256* cT2 = nullptr;
    if (coset_gen) {
        if (ds.coset_cap < (size_t)t1n + t2n) {
            if (ds.d_cosetA) (void)hipFree(ds.d_cosetA);
            HIP_TRY(hipMalloc(&ds.d_cosetA, ((size_t)t1n + t2n) * sizeof(fp256)));
            ds.coset_cap = (size_t)t1n + t2n;
        }
        cT1 = ds.d_cosetA;
        cT2 = ds.d_cosetA + t1n;
        fp256 gT2;
        host_pow_u32(gT2, *coset_gen, 1u << TW_LOW_BITS);
        build_tables(ds, *coset_gen, cT1, t1n);
        build_tables(ds, gT2, cT2, t2n);
    }
    // n^{-1} scale for inverse
    fp256 scale;
    ff_set_one<Fr>(scale);
    if (inverse) {
        fp256 ncanon, nm;
        ff_set_zero(ncanon);
        ncanon.l[log_n >> 5] = 1u << (log_n & 31);
        ff_to_mont<Fr>(nm, ncanon);
        ff_inv<Fr>(scale, nm);
    }
    hipStream_t st = ds.stream;
    // Tile thread count: L/4 (radix-4 butterflies occupy exactly L/4
    // threads; smaller blocks co-reside per CU, LDS permitting, keeping
    // occupancy), floored at 128 for tiny tiles where 64-thread blocks
    // underutilize. Measured (r2 TDIV sweep): 2^22 coset 1.45 -> 1.19 ms,
    // 2^23 fwd 2.54 -> 2.26, 2^24 unchanged, 2^20 within 4% of its best.
    // SPECTRE_NTT_TDIV=1|2|4 overrides for A/B.
    static const uint32_t kTDiv = []() {
        const char* e = getenv("SPECTRE_NTT_TDIV");
        int v = e ? atoi(e) : 0;
        return (uint32_t)(v == 1 || v == 2 || v == 4 ? v : 0);  // 0 = auto
    }();
    auto tdiv = [](uint32_t L) {
        uint32_t t = kTDiv ? L / kTDiv : (L / 4 > 128 ? L / 4 : 128);
        return t < 64 ? 64u : (t > NTT_THREADS ? (uint32_t)NTT_THREADS : t);
    };
    const fp256* fwd_cT1 = (coset_gen && !inverse) ? cT1 : nullptr;
    const fp256* fwd_cT2 = (coset_gen && !inverse) ? cT2 : nullptr;
    const fp256* inv_cT1 = (coset_gen && inverse) ? cT1 : nullptr;
    const fp256* inv_cT2 = (coset_gen && inverse) ? cT2 : nullptr;
    if (plan->k3 > 0) {
        // three-pass path (log_n > 24, or SPECTRE_NTT_FORCE3 for testing)
        const uint32_t kA = plan->k1, kB = plan->k2, kC = plan->k3;
        const uint32_t A = 1u << kA, B = 1u << kB, C = 1u << kC;
        fp256 one;
        ff_set_one<Fr>(one);
        const fp256* T1 = plan->twB;
        const fp256* T2 = plan->twB + t1n;
        auto tdim = tdiv;
        // radix-4 instantiation only when every thread gets a quad
        auto axis = [](uint32_t L, uint32_t threads) {
            return (!kForceRadix2 && L >= 4 * threads) ? k_ntt_axis<true>
                                                       : k_ntt_axis<false>;
        };
        hipLaunchKernelGGL(axis(A, tdim(A)), dim3(B * C), dim3(tdim(A)),
                           LDS_BYTES(A), st, d_data, ds.d_ntt_tmp, plan->tw1,
                           T1, T2, fwd_cT1, fwd_cT2, /*coset_on_load=*/1, one,
                           0, kA, kB + kC, /*post_mode=*/1, 0, 0, 0, 0, 0);
        hipLaunchKernelGGL(axis(B, tdim(B)), dim3(A * C), dim3(tdim(B)),
                           LDS_BYTES(B), st, ds.d_ntt_tmp, ds.d_ntt_tmp,
                           plan->tw2, T1, T2, nullptr, nullptr, 0, one, 0, kB,
                           kC, /*post_mode=*/2, kA, kC, 0, 0, 0);
        hipLaunchKernelGGL(axis(C, tdim(C)), dim3(A * B), dim3(tdim(C)),
                           LDS_BYTES(C), st, ds.d_ntt_tmp, d_data, plan->tw3,
                           T1, T2, inv_cT1, inv_cT2, /*coset_on_load=*/0,
                           scale, inverse ? 1 : 0, kC, 0, /*post_mode=*/0, 0,
                           0, /*final_scatter=*/1, A, B);
    } else if (plan->k1 > 0) {
        const uint32_t tc = tdiv(n1);
        const uint32_t tr = tdiv(n2);
        hipLaunchKernelGGL((!kForceRadix2 && n1 >= 4 * tc) ? k_ntt_col<true>
                                                           : k_ntt_col<false>,
                           dim3(n2), dim3(tc),
                           LDS_BYTES(n1), st, d_data,
                           ds.d_ntt_tmp, plan->tw1, plan->twB,
                           plan->twB + t1n, fwd_cT1, fwd_cT2, plan->k1,
                           plan->k2);
        hipLaunchKernelGGL((!kForceRadix2 && n2 >= 4 * tr) ? k_ntt_row<true>
                                                           : k_ntt_row<false>,
                           dim3(n1), dim3(tr),
                           LDS_BYTES(n2), st, ds.d_ntt_tmp,
                           d_data, plan->tw2, inv_cT1, inv_cT2,
                           /*coset_on_load=*/0, scale, inverse ? 1 : 0,
                           plan->k1, plan->k2);
    } else {
        // single pass; forward coset applies on load, inverse on store
        const fp256* cc1 = coset_gen ? cT1 : nullptr;
        const fp256* cc2 = coset_gen ? cT2 : nullptr;
        const uint32_t t1p = tdiv(n2);
        hipLaunchKernelGGL((!kForceRadix2 && n2 >= 4 * t1p) ? k_ntt_row<true>
                                                            : k_ntt_row<false>,
                           dim3(1), dim3(t1p),
                           LDS_BYTES(n2), st, d_data, d_data,
                           plan->tw2, cc1, cc2,
                           /*coset_on_load=*/inverse ? 0 : 1, scale,
                           inverse ? 1 : 0, 0, plan->k2);
    }
    HIP_TRY(hipStreamSynchronize(st));
    HIP_TRY(hipGetLastError());
    return 0;
}
