// msm.hip — BN254 G1 multi-scalar multiplication (Pippenger) for gfx950.
//
// Computes what halo2curves-axiom 0.5.2 `best_multiexp` computes (the value
// Sum_i k_i * P_i; windowing is free — see oracle/bn254.c header for the
// parity contract). The whole pipeline is batch-generic: `nbatch` scalar
// vectors over ONE shared base set run as a single sort/accumulate pass
// (create_proof commits ~17 advice columns against the same SRS back to
// back — SURVEY.md §8f-2); nbatch = 1 is the plain best_multiexp call.
//
// MI355X-native structure:
//   1. k_msm_digits     — one thread per point index: optional
//                         Montgomery->canonical conversion in registers,
//                         signed MSM_WBITS-bit window recoding, emit
//                         (bucket key, point index|sign) pairs for every
//                         (batch, window). Writes are (batch,window)-major,
//                         so each slice is written coalesced.
//   2. radix sort       — hipCUB DeviceRadixSort over batch*window*bucket
//                         keys; zero digits get a sentinel key sorting last.
//   3. k_bucket_offsets — binary-search segment bounds per bucket.
//   4. k_bucket_acc     — equal-work partitioning: each thread owns exactly
//                         MSM_ACC_E sorted entries (serial Jacobian+affine
//                         mixed adds); interior runs write their bucket
//                         exclusively, boundary runs are merged by
//                         k_bucket_fix. Deterministic by construction, and
//                         the affine result is canonical, so ANY schedule
//                         yields bit-identical output bytes.
//   5. k_window_chunks  — per MSM_CHUNK consecutive buckets: weighted suffix
//                         sum + double-and-add lift to the window offset.
//   6. k_window_sum     — one block per (batch, window), pairwise LDS tree
//                         to the final window sums in a single launch.
// Host side (ffi.cpp) finishes each batch with the window Horner (~255
// doublings + adds) and one field inversion to affine.
//
// All field math is 8x32-limb Montgomery CIOS (ff.hpp) — VALU integer work
// (measured at the chip's v_mad_u64_u32 issue ceiling); MFMA does not apply
// to modular arithmetic.
#include "internal.hpp"
#include <hipcub/hipcub.hpp>

#define THREADS 256
// Point-arithmetic kernels hold a 24-VGPR Jacobian accumulator plus formula
// temporaries; at the default 4-waves/SIMD register budget (128 VGPR) hipcc
// spills to scratch. Allowing 2 waves/SIMD (256 VGPR) removes the spills —
// latency hiding comes from the serial-add structure, not occupancy
// (measured: 1 add-chain/thread already saturates the VALU issue pipe).
#define PT_KERNEL __global__ __launch_bounds__(THREADS, 2)
// The accumulate/fix kernels have no LDS or barriers, so their block size
// only sets the residency granule. MEASURED (r2): 64/128/256-thread blocks
// are within run-to-run noise (acc 2.26-2.29 ms) — the 24% memory-parked
// wave time (SQ_WAIT_ANY) does not yield to the extra wave 128-thread
// blocks admit (84 VGPRs -> 5 waves/SIMD vs 4). Kept tunable.
#ifndef MSM_ACC_THREADS
#define MSM_ACC_THREADS 256
#endif
#define ACC_KERNEL __global__ __launch_bounds__(MSM_ACC_THREADS, 2)

// ---- kernel 1: signed window decomposition --------------------------------
// w_lo/w_cnt: emit only windows [w_lo, w_lo+w_cnt) — window-sharded
// multi-GPU (each rank owns disjoint windows over ALL points, so bucket
// work AND the reduction tail divide by the rank count; the exchange is a
// pure allgather of disjoint window sums). The signed recoding carry runs
// over ALL windows regardless (it propagates from window 0 upward).
__global__ void k_msm_digits(const uint8_t* __restrict__ scalars, uint64_t n,
                             uint32_t nbatch, int canonical, uint32_t w_lo,
                             uint32_t w_cnt,
                             uint32_t* __restrict__ keys,
                             uint32_t* __restrict__ vals) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const uint32_t nb_local = w_cnt * MSM_BPW;
    const uint32_t skip_key = nbatch * nb_local;
    for (uint32_t b = 0; b < nbatch; b++) {
        fp256 s;
        ff_from_bytes(s, scalars + 32 * (b * n + i));
        if (!canonical) ff_from_mont<Fr>(s, s);
        uint32_t carry = 0;
        for (int w = 0; w < MSM_NWIN; w++) {
            // c-bit window w of the canonical scalar (may cross limbs)
            const uint32_t bit0 = (uint32_t)w * MSM_WBITS;
            const uint32_t li = bit0 >> 5, sh = bit0 & 31;
            uint64_t pair = (uint64_t)s.l[li];
            if (li < 7) pair |= (uint64_t)s.l[li + 1] << 32;
            uint32_t d =
                ((uint32_t)(pair >> sh) & ((1u << MSM_WBITS) - 1)) + carry;
            const uint32_t half = 1u << (MSM_WBITS - 1);
            const uint32_t full = 1u << MSM_WBITS;
            uint32_t key, val = (uint32_t)i;
            if (d == 0) {
                carry = 0;
                key = skip_key;
            } else if (d <= half) {  // positive digit, magnitude d
                carry = 0;
                key = b * nb_local + ((uint32_t)w - w_lo) * MSM_BPW + (d - 1);
            } else if (d == full) {  // max digit + carry: digit 0, carry out
                carry = 1;
                key = skip_key;
            } else {  // negative digit, magnitude 2^c - d
                carry = 1;
                key = b * nb_local + ((uint32_t)w - w_lo) * MSM_BPW +
                      (full - d - 1);
                val |= 0x80000000u;
            }
            if ((uint32_t)w < w_lo || (uint32_t)w >= w_lo + w_cnt) continue;
            const uint64_t slot =
                ((uint64_t)b * w_cnt + ((uint32_t)w - w_lo)) * n + i;
            keys[slot] = key;
            vals[slot] = val;
        }
    }
}

// ---- kernel 3: per-bucket segment bounds via binary search ----------------
__global__ void k_bucket_offsets(const uint32_t* __restrict__ keys,
                                 uint64_t nent, uint32_t nb_total,
                                 uint32_t* __restrict__ off) {
    uint32_t b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b > nb_total) return;
    uint64_t lo = 0, hi = nent;
    while (lo < hi) {
        uint64_t mid = (lo + hi) >> 1;
        if (keys[mid] < b) lo = mid + 1;
        else hi = mid;
    }
    off[b] = (uint32_t)lo;
}

// ---- kernel 4: bucket accumulation, equal-work partitioning ---------------
// Bucket sizes are Poisson(mean ent/NB); one-thread-per-bucket makes the
// busiest lane in a wave ~1.5x the mean (SIMT runs at the max). Instead each
// thread owns exactly MSM_ACC_E consecutive SORTED entries: runs that start
// AND end strictly inside the range write their bucket directly (exclusive);
// the first and last (potentially thread-spanning) runs go to side arrays
// keyed by bucket, merged by k_bucket_fix.
ACC_KERNEL void k_bucket_acc(const uint32_t* __restrict__ off,
                            const uint32_t* __restrict__ keys,
                            const uint32_t* __restrict__ vals,
                            const g1_affine* __restrict__ bases,
                            uint32_t nb_total, g1_jac* __restrict__ buckets,
                            uint32_t* __restrict__ firstK,
                            g1_jac* __restrict__ firstP,
                            uint32_t* __restrict__ lastK,
                            g1_jac* __restrict__ lastP) {
    const uint32_t ent = off[nb_total];  // real (non-skip) entries
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t lo = (uint64_t)t * MSM_ACC_E;
    if (lo >= ent) return;
    uint64_t hi = lo + MSM_ACC_E;
    if (hi > ent) hi = ent;
    uint32_t k = keys[lo];
    // MEASURED NEGATIVE RESULT (r2): accumulating runs in XYZZ
    // (mADD-2008-s, 8M+2S — see g1.hpp) with a per-run Jacobian conversion
    // is bit-exact but SLOWER: 152 VGPRs / 3 waves/SIMD vs the Jacobian
    // accumulator's 84 / 4-5, and bucket_acc went 2.31 -> 3.09 ms at
    // n=2^20 — the occupancy loss outweighs the ~9% fewer multiplies.
    g1_jac acc;
    g1j_set_inf(acc);
    bool first = true;
    for (uint64_t j = lo; j < hi; j++) {
        uint32_t kj = keys[j];
        if (kj != k) {  // flush completed run
            if (first) {
                firstK[t] = k;
                firstP[t] = acc;
                first = false;
            } else if (k < nb_total) {
                buckets[k] = acc;  // interior run: exclusive writer
            }
            g1j_set_inf(acc);
            k = kj;
        }
        if (kj < nb_total) {
            uint32_t v = vals[j];
            // plain (cached) loads: bases are re-read by all 16 windows, so
            // L1/L2 residency pays (non-temporal loads measured 5% slower)
            g1_affine p = bases[v & 0x7fffffffu];
            if (v & 0x80000000u) ff_neg<Fq>(p.y, p.y);
            g1j_madd_ip(acc, p);
        }
    }
    if (first) {  // whole range is one run
        firstK[t] = k;
        firstP[t] = acc;
        lastK[t] = 0xffffffffu;
    } else {
        lastK[t] = k;
        lastP[t] = acc;
    }
}

// merge boundary partials: bucket b's segment [s,e) spans threads ts..te;
// interior-only buckets were already written by their exclusive thread.
ACC_KERNEL void k_bucket_fix(const uint32_t* __restrict__ off,
                            const uint32_t* __restrict__ firstK,
                            const g1_jac* __restrict__ firstP,
                            const uint32_t* __restrict__ lastK,
                            const g1_jac* __restrict__ lastP,
                            uint32_t nb_total, g1_jac* __restrict__ buckets) {
    uint32_t b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= nb_total) return;
    uint32_t s = off[b], e = off[b + 1];
    if (s == e) {
        g1j_set_inf(buckets[b]);
        return;
    }
    uint32_t ts = s / MSM_ACC_E, te = (e - 1) / MSM_ACC_E;
    // interior (already written by its exclusive thread) iff the run starts
    // after the thread's range start AND ends before the thread's range end —
    // note a run ending at the END OF DATA (e == ent) ended the thread's
    // loop, so it lives in lastP, not in buckets[].
    const uint32_t ent = off[nb_total];
    if (ts == te && (s % MSM_ACC_E) && (e % MSM_ACC_E) && e != ent)
        return;
    g1_jac acc;
    g1j_set_inf(acc);
    for (uint32_t t = ts; t <= te; t++) {  // ascending = sorted entry order
        if (firstK[t] == b) g1j_add_ip(acc, firstP[t]);
        if (lastK[t] == b) g1j_add_ip(acc, lastP[t]);
    }
    buckets[b] = acc;
}

// ---- kernel 5: weighted chunk reduction -----------------------------------
// Window-local bucket j has multiplier (j+1). For the chunk covering
// window-local buckets [m, m+CHUNK):
//   sum_j (j+1) B[m+j]  (suffix running sums)  +  m * sum_j B[m+j]
// (double-and-add; m < 2^(WBITS-1)). Uniform in the flattened
// (batch*NWIN + w) window index, so batching needs no changes here.
PT_KERNEL void k_window_chunks(const g1_jac* __restrict__ buckets,
                               uint32_t total_chunks,
                               g1_jac* __restrict__ out) {
    const uint32_t nchunks_pw = MSM_BPW / MSM_CHUNK;
    uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= total_chunks) return;
    uint32_t cw = t % nchunks_pw;
    const g1_jac* B = buckets + (uint64_t)t * MSM_CHUNK;
    // Phase order keeps at most TWO Jacobian accumulators live (3 would push
    // past 256 VGPRs and spill): accW is parked in `out` before the
    // double-and-add, then re-loaded for the final add.
    g1_jac accT, accW;
    g1j_set_inf(accT);
    g1j_set_inf(accW);
    for (int j = MSM_CHUNK - 1; j >= 0; j--) {
        g1j_add_ip(accT, B[j]);
        g1j_add_ip(accW, accT);
    }
    out[t] = accW;  // park; accW dead
    const uint32_t m = cw * MSM_CHUNK;
    if (m && !g1j_is_inf(accT)) {
        g1_jac acc;
        g1j_set_inf(acc);
        for (int bit = MSM_WBITS - 1; bit >= 0; bit--) {
            g1j_dbl_ip(acc);
            if ((m >> bit) & 1) g1j_add_ip(acc, accT);
        }
        g1_jac w = out[t];  // accT dead; (w, acc) live
        g1j_add_ip(w, acc);
        out[t] = w;
    }
}

// ---- kernel 6: per-window LDS-tree sum ------------------------------------
// One block per flattened (batch, window) sums its BPW/CHUNK chunk points:
// threads grid-stride their share, then a pairwise LDS tree — serial depth
// ~16 adds in ONE launch (a 3-level cascade of tiny grids was ~0.8 ms of
// launch+latency overhead).
#define WSUM_THREADS 256
__global__ __launch_bounds__(WSUM_THREADS, 2) void k_window_sum(
    const g1_jac* __restrict__ in, g1_jac* __restrict__ out) {
    __shared__ g1_jac lds[WSUM_THREADS / 2];
    const uint32_t per_win = MSM_BPW / MSM_CHUNK;
    const uint32_t w = blockIdx.x;
    const uint32_t t = threadIdx.x;
    g1_jac acc;
    g1j_set_inf(acc);
    for (uint32_t j = t; j < per_win; j += blockDim.x)
        g1j_add_ip(acc, in[(uint64_t)w * per_win + j]);
    for (uint32_t k = WSUM_THREADS / 2; k >= 1; k >>= 1) {
        if (t >= k && t < 2 * k) lds[t - k] = acc;
        __syncthreads();
        if (t < k) g1j_add_ip(acc, lds[t]);
        __syncthreads();
    }
    if (t == 0) out[w] = acc;
}

// ---- host orchestration ---------------------------------------------------
static int ensure_msm_scratch(MsmSlot& ds, hipStream_t stream, uint64_t n,
                              uint32_t nbatch, uint32_t w_cnt) {
    const uint64_t ent = (uint64_t)nbatch * w_cnt * n;
    const uint64_t nbt = (uint64_t)nbatch * w_cnt * MSM_BPW;
    if (ds.ent_cap < ent) {
        // free-and-null before reallocating: a failed hipMalloc below must
        // not leave dangling pointers with a stale capacity (the caps are
        // only advanced after every allocation in the group succeeds).
        ds.ent_cap = 0;
        for (void** p : {(void**)&ds.d_keys_in, (void**)&ds.d_keys_out,
                         (void**)&ds.d_vals_in, (void**)&ds.d_vals_out,
                         (void**)&ds.d_firstK, (void**)&ds.d_lastK,
                         (void**)&ds.d_firstP, (void**)&ds.d_lastP})
            if (*p) { (void)hipFree(*p); *p = nullptr; }
        HIP_TRY(hipMalloc(&ds.d_keys_in, ent * 4));
        HIP_TRY(hipMalloc(&ds.d_keys_out, ent * 4));
        HIP_TRY(hipMalloc(&ds.d_vals_in, ent * 4));
        HIP_TRY(hipMalloc(&ds.d_vals_out, ent * 4));
        const uint64_t nt = (ent + MSM_ACC_E - 1) / MSM_ACC_E;
        HIP_TRY(hipMalloc(&ds.d_firstK, nt * 4));
        HIP_TRY(hipMalloc(&ds.d_lastK, nt * 4));
        HIP_TRY(hipMalloc(&ds.d_firstP, nt * sizeof(g1_jac)));
        HIP_TRY(hipMalloc(&ds.d_lastP, nt * sizeof(g1_jac)));
        ds.ent_cap = ent;
    }
    if (ds.nb_cap < nbt) {
        ds.nb_cap = 0;
        for (void** p : {(void**)&ds.d_offsets, (void**)&ds.d_buckets,
                         (void**)&ds.d_red})
            if (*p) { (void)hipFree(*p); *p = nullptr; }
        HIP_TRY(hipMalloc(&ds.d_offsets, (nbt + 1) * 4));
        HIP_TRY(hipMalloc(&ds.d_buckets, nbt * sizeof(g1_jac)));
        HIP_TRY(hipMalloc(&ds.d_red,
                          (nbt / MSM_CHUNK + nbatch * w_cnt) * sizeof(g1_jac)));
        ds.nb_cap = nbt;
    }
    size_t sort_need = 0;
    (void)hipcub::DeviceRadixSort::SortPairs(nullptr, sort_need, ds.d_keys_in,
                                             ds.d_keys_out, ds.d_vals_in,
                                             ds.d_vals_out, (int64_t)ent, 0,
                                             32, stream);
    if (ds.sort_tmp_cap < sort_need) {
        if (ds.d_sort_tmp) (void)hipFree(ds.d_sort_tmp);
        HIP_TRY(hipMalloc(&ds.d_sort_tmp, sort_need));
        ds.sort_tmp_cap = sort_need;
    }
    return 0;
}

static int end_bit_for(uint64_t max_key) {
    int b = 0;
    while ((1ull << b) <= max_key) b++;
    return b;
}

int msm_batch_shard_device(spectre_gpu_ctx* ctx, int dev,
                           const g1_affine* d_bases, const uint8_t* d_scalars,
                           uint32_t nbatch, uint64_t n, uint32_t flags,
                           g1_jac* winsums_host, double* stage_ms, bool sync,
                           int slot) {
    return msm_batch_windows_device(ctx, dev, d_bases, d_scalars, nbatch, n,
                                    flags, 0, MSM_NWIN, winsums_host,
                                    stage_ms, sync, slot);
}

// window-ranged pipeline: nwin = w_cnt windows starting at w_lo; winsums_host
// receives nbatch * w_cnt Jacobian sums (full-range callers pass 0, NWIN).
int msm_batch_windows_device(spectre_gpu_ctx* ctx, int dev,
                             const g1_affine* d_bases,
                             const uint8_t* d_scalars, uint32_t nbatch,
                             uint64_t n, uint32_t flags, uint32_t w_lo,
                             uint32_t w_cnt, g1_jac* winsums_host,
                             double* stage_ms, bool sync, int slot) {
    DeviceState& dstate = ctx->devs[dev];
    HIP_TRY(hipSetDevice(dstate.device_id));
    if (slot < 0 || slot > 2) {
        set_err("msm: slot %d out of range", slot);
        return -3;
    }
    MsmSlot& ds = dstate.slots[slot];
    if (!sync && ds.pending_dst) {
        // a second in-flight call on one slot would overwrite the pending
        // D2H silently — fail loudly instead (callers must slot_wait first)
        set_err("msm: slot %d already has a call in flight", slot);
        return -3;
    }
    if (!ds.stream) {
        if (slot == 0) ds.stream = dstate.stream;
        else HIP_TRY(hipStreamCreate(&ds.stream));
    }
    if (nbatch == 0 || nbatch > SPECTRE_MSM_MAX_BATCH) {
        set_err("msm: nbatch %u out of range [1,%d]", nbatch,
                SPECTRE_MSM_MAX_BATCH);
        return -3;
    }
    // Sort entries and segment offsets are indexed with uint32: at exactly
    // nbatch*NWIN*n == 2^32 the final offset off[nb_total] wraps to 0 and the
    // result is silently wrong. Reject; callers split such jobs (e.g. batch
    // 32 x 2^23 -> 2 x 16).

    if (w_lo >= MSM_NWIN || w_cnt == 0 || w_lo + w_cnt > MSM_NWIN) {
        set_err("msm: window range [%u, %u+%u) out of [0,%d)", w_lo, w_lo,
                w_cnt, MSM_NWIN);
        return -3;
    }
    if ((uint64_t)nbatch * w_cnt * n >= (1ull << 32)) {
        set_err("msm: nbatch*%u*n = %llu exceeds 2^32 sort-entry limit "
                "(split the batch or shard the MSM)",
                w_cnt, (unsigned long long)((uint64_t)nbatch * w_cnt * n));
        return -3;
    }
    if (n == 0) {
        for (uint32_t w = 0; w < nbatch * w_cnt; w++)
            g1j_set_inf(winsums_host[w]);
        return 0;
    }
    int rc = ensure_msm_scratch(ds, ds.stream, n, nbatch, w_cnt);
    if (rc) return rc;
    const uint64_t ent = (uint64_t)nbatch * w_cnt * n;
    const uint32_t nbt = nbatch * w_cnt * MSM_BPW;
    const int canonical = (flags & SPECTRE_SCALARS_CANONICAL) ? 1 : 0;
    hipStream_t st = ds.stream;

    // optional per-stage HIP events (on the library stream — this is the
    // live in-bench measurement the roofline report is built from)
    hipEvent_t ev[7];
    if (stage_ms)
        for (auto& e : ev) HIP_TRY(hipEventCreate(&e));
#define STAMP(i) \
    if (stage_ms) HIP_TRY(hipEventRecord(ev[i], st));

    STAMP(0);
    uint32_t blocks = (uint32_t)((n + THREADS - 1) / THREADS);
    hipLaunchKernelGGL(k_msm_digits, dim3(blocks), dim3(THREADS), 0, st,
                       d_scalars, n, nbatch, canonical, w_lo, w_cnt,
                       ds.d_keys_in, ds.d_vals_in);
    STAMP(1);
    size_t tmp = ds.sort_tmp_cap;
    (void)hipcub::DeviceRadixSort::SortPairs(
        ds.d_sort_tmp, tmp, ds.d_keys_in, ds.d_keys_out, ds.d_vals_in,
        ds.d_vals_out, (int64_t)ent, 0, end_bit_for(nbt), st);
    STAMP(2);
    hipLaunchKernelGGL(k_bucket_offsets,
                       dim3((nbt + 1 + THREADS - 1) / THREADS), dim3(THREADS),
                       0, st, ds.d_keys_out, ent, nbt, ds.d_offsets);
    STAMP(3);
    const uint32_t nt_acc = (uint32_t)((ent + MSM_ACC_E - 1) / MSM_ACC_E);
    hipLaunchKernelGGL(k_bucket_acc,
                       dim3((nt_acc + MSM_ACC_THREADS - 1) / MSM_ACC_THREADS),
                       dim3(MSM_ACC_THREADS), 0, st, ds.d_offsets,
                       ds.d_keys_out, ds.d_vals_out, d_bases, nbt,
                       ds.d_buckets, ds.d_firstK, ds.d_firstP, ds.d_lastK,
                       ds.d_lastP);
    hipLaunchKernelGGL(k_bucket_fix,
                       dim3((nbt + MSM_ACC_THREADS - 1) / MSM_ACC_THREADS),
                       dim3(MSM_ACC_THREADS), 0, st, ds.d_offsets,
                       ds.d_firstK, ds.d_firstP, ds.d_lastK, ds.d_lastP, nbt,
                       ds.d_buckets);
    STAMP(4);
    const uint32_t nchunks = nbt / MSM_CHUNK;
    g1_jac* red0 = ds.d_red;
    g1_jac* red1 = ds.d_red + nchunks;
    hipLaunchKernelGGL(k_window_chunks,
                       dim3((nchunks + THREADS - 1) / THREADS), dim3(THREADS),
                       0, st, ds.d_buckets, nchunks, red0);
    STAMP(5);
    hipLaunchKernelGGL(k_window_sum, dim3(nbatch * w_cnt),
                       dim3(WSUM_THREADS), 0, st, red0, red1);
    STAMP(6);
    const uint32_t nw = nbatch * w_cnt;
    if (ds.h_wins_cap < nw) {
        if (ds.h_wins) (void)hipHostFree(ds.h_wins);
        ds.h_wins_cap = 0;
        HIP_TRY(hipHostMalloc((void**)&ds.h_wins, (size_t)nw * sizeof(g1_jac)));
        ds.h_wins_cap = nw;
    }
    HIP_TRY(hipMemcpyAsync(ds.h_wins, red1, (size_t)nw * sizeof(g1_jac),
                           hipMemcpyDeviceToHost, st));
    uint32_t ent_real = 0;
    if (stage_ms)
        HIP_TRY(hipMemcpyAsync(&ent_real, ds.d_offsets + nbt, 4,
                               hipMemcpyDeviceToHost, st));
    if (!sync && !stage_ms) {
        HIP_TRY(hipGetLastError());
        ds.pending_dst = winsums_host;  // drained by msm_slot_drain
        ds.pending_n = nw;
        return 0;
    }
    HIP_TRY(hipStreamSynchronize(st));
    HIP_TRY(hipGetLastError());
    memcpy(winsums_host, ds.h_wins, (size_t)nw * sizeof(g1_jac));
    if (stage_ms) {
        float ms;
        for (int i = 0; i < 6; i++) {
            HIP_TRY(hipEventElapsedTime(&ms, ev[i], ev[i + 1]));
            stage_ms[i] = ms;
        }
        HIP_TRY(hipEventElapsedTime(&ms, ev[0], ev[6]));
        stage_ms[6] = ms;
        stage_ms[7] = (double)ent_real;
        for (auto& e : ev) (void)hipEventDestroy(e);
    }
#undef STAMP
    return 0;
}

int msm_shard_device(spectre_gpu_ctx* ctx, int dev, const g1_affine* d_bases,
                     const uint8_t* d_scalars, uint64_t n, uint32_t flags,
                     g1_jac* winsums_host, double* stage_ms) {
    return msm_batch_shard_device(ctx, dev, d_bases, d_scalars, 1, n, flags,
                                  winsums_host, stage_ms);
}

// Synchronize a slot's stream and deliver the pending window sums (pinned ->
// caller memory). Idempotent when nothing is pending.
int msm_slot_drain(spectre_gpu_ctx* ctx, int dev, int slot) {
    DeviceState& dstate = ctx->devs[dev];
    if (slot < 0 || slot > 2) {
        set_err("slot_drain: slot %d out of range", slot);
        return -1;
    }
    MsmSlot& sl = dstate.slots[slot];
    HIP_TRY(hipSetDevice(dstate.device_id));
    if (sl.stream) HIP_TRY(hipStreamSynchronize(sl.stream));
    HIP_TRY(hipGetLastError());
    if (sl.pending_dst) {
        memcpy(sl.pending_dst, sl.h_wins, (size_t)sl.pending_n * sizeof(g1_jac));
        sl.pending_dst = nullptr;
        sl.pending_n = 0;
    }
    return 0;
}
