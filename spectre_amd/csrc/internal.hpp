// internal.hpp — shared internals of libspectre_gpu.so (not installed).
#pragma once
#include <hip/hip_runtime.h>
#include "spectre_gpu.h"
#include "ff.hpp"
#include "g1.hpp"
#include <array>
#include <map>
#include <mutex>
#include <string>
#include <vector>

// MSM window configuration: MSM_NWIN signed MSM_WBITS-bit windows covering
// >= 255 bits. Signed-digit recoding halves the bucket count: digit
// magnitudes in [1, 2^(WBITS-1)], bucket index = magnitude-1, negative
// digits negate the point (one Fq negation). Canonical BN254 Fr scalars are
// < 2^254 and NWIN*WBITS >= 255, so the top window never recodes negative
// and never carries out. Measured on MI355X: c=15 trades +8% bucket work
// for a 2x smaller tail and nets out slightly WORSE than c=16 (5.14 vs
// 5.02 ms at n=2^20) - c=16 kept.
#define MSM_WBITS 16
#define MSM_NWIN 16                        // ceil(255 / WBITS)
#define MSM_BPW (1u << (MSM_WBITS - 1))    // buckets per window
#define MSM_NB_TOTAL (MSM_NWIN * MSM_BPW)
#define MSM_SKIP_KEY MSM_NB_TOTAL          // sorts after all real keys
#define MSM_SORT_BITS 20                   // key range < 2^20
#ifndef MSM_CHUNK
#define MSM_CHUNK 8                        // buckets per reduction thread
#endif
#ifndef MSM_ACC_E
#define MSM_ACC_E 64                       // sorted entries per acc thread
#endif

struct NttPlan {
    fp256* tw1 = nullptr;  // butterfly twiddles, axis 1
    fp256* tw2 = nullptr;  // butterfly twiddles, axis 2
    fp256* tw3 = nullptr;  // butterfly twiddles, axis 3 (3-pass only)
    fp256* twB = nullptr;  // T1 || T2 omega-power lookup (w^e = T1[e&fff]*T2[e>>12])
    uint32_t k1 = 0, k2 = 0, k3 = 0;
};

struct CachedBases {
    g1_affine* d_ptr = nullptr;
    uint64_t n = 0;
};

// Per-slot MSM pipeline state: two slots per device let call i+1's
// digits/sort/accumulate fill the machine while call i's latency-bound
// reduction tail (~1.3 ms at <=1 wave/SIMD) drains on the other stream —
// back-to-back commits (create_proof issues ~45 per proof) overlap instead
// of serializing on the tail.
struct MsmSlot {
    hipStream_t stream = nullptr;  // slot 0 aliases DeviceState::stream
    uint32_t* d_keys_in = nullptr;
    uint32_t* d_keys_out = nullptr;
    uint32_t* d_vals_in = nullptr;
    uint32_t* d_vals_out = nullptr;
    size_t ent_cap = 0;  // capacity in entries (16*n)
    void* d_sort_tmp = nullptr;
    size_t sort_tmp_cap = 0;
    size_t nb_cap = 0;              // bucket-array capacity (batched)
    uint32_t* d_offsets = nullptr;  // nb_cap + 1
    g1_jac* d_buckets = nullptr;    // MSM_NB_TOTAL
    uint32_t* d_firstK = nullptr;   // boundary-run side arrays (ent_cap/ACC_E)
    uint32_t* d_lastK = nullptr;
    g1_jac* d_firstP = nullptr;
    g1_jac* d_lastP = nullptr;
    g1_jac* d_red = nullptr;        // reduction ping-pong (NB_TOTAL/CHUNK * 2)
    // Window sums come back through a PINNED per-slot buffer: an async D2H
    // into caller (pageable) memory silently blocks the calling thread on
    // ROCm, which serialized the whole two-slot pipeline in r2's first
    // measurement. The wait/drain step memcpys pinned -> caller.
    g1_jac* h_wins = nullptr;   // pinned, h_wins_cap elements
    size_t h_wins_cap = 0;
    g1_jac* pending_dst = nullptr;  // caller buffer for the in-flight call
    uint32_t pending_n = 0;         // elements pending (nbatch * NWIN)
};

struct DeviceState {
    int device_id = 0;
    hipStream_t stream = nullptr;
    MsmSlot slots[3];  // >=2 enables the async pipeline; 3rd for depth-3 A/B
    int next_slot = 0;  // round-robin for the async API
    uint8_t* d_scalars = nullptr;
    size_t scal_cap = 0;  // bytes
    g1_affine* d_bases = nullptr;
    size_t base_cap = 0;  // points
    // key = (bases_id, n, shard layout): layout is num_gpus for the sharded
    // path (this device's contiguous chunk of an n-point set split num_gpus
    // ways) and 1 for the full set (batch path == num_gpus=1 shard).
    std::map<std::array<uint64_t, 3>, CachedBases> bases_cache;
    // ---- pinned staging for host-pointer uploads/downloads ----
    uint8_t* h_stage[2] = {nullptr, nullptr};
    hipEvent_t stage_ev[2] = {nullptr, nullptr};
    // ---- NTT scratch ----
    fp256* d_ntt_tmp = nullptr;
    size_t ntt_cap = 0;  // elements
    fp256* d_ntt_io = nullptr;  // cached device buffer for host-pointer NTTs
    size_t ntt_io_cap = 0;      // elements (the Rust seam calls this path
                                // ~40-60x per proof; no per-call hipMalloc)
    fp256* d_cosetA = nullptr;  // coset base tables (n2 resp. n1 entries)
    fp256* d_cosetB = nullptr;
    size_t coset_cap = 0;
    // ---- gate-eval scratch (program, constants, column-pointer array) ----
    uint8_t* d_gate = nullptr;
    size_t gate_cap = 0;  // bytes
    std::map<std::array<uint8_t, 40>, NttPlan> plans;  // omega||log_n||inverse
};

struct spectre_gpu_ctx {
    std::vector<DeviceState> devs;
    std::recursive_mutex mu;
};

void set_err(const char* fmt, ...);

#define HIP_TRY(x)                                                        \
    do {                                                                  \
        hipError_t _e = (x);                                              \
        if (_e != hipSuccess) {                                           \
            set_err("%s:%d %s: %s", __FILE__, __LINE__, #x,               \
                    hipGetErrorString(_e));                               \
            return -2;                                                    \
        }                                                                 \
    } while (0)

// msm.hip — runs the full Pippenger pipeline for one shard on device `dev`'s
// stream and writes the MSM_NWIN Jacobian window sums to host memory
// (synchronizes the stream). If stage_ms != nullptr, per-stage HIP-event
// timings are written: [0]=digits [1]=sort [2]=offsets [3]=bucket_acc
// [4]=chunks [5]=reduce [6]=total-gpu [7]=real (non-zero-digit) entry count.
int msm_shard_device(spectre_gpu_ctx* ctx, int dev, const g1_affine* d_bases,
                     const uint8_t* d_scalars, uint64_t n, uint32_t flags,
                     g1_jac* winsums_host, double* stage_ms = nullptr);
// batch variant: nbatch scalar vectors (batch-major, nbatch*n*32 B) over one
// shared base set; winsums_host receives nbatch*MSM_NWIN Jacobian sums.
// sync=false: enqueue the pipeline + async D2H of winsums on the slot's
// stream and return without synchronizing (multi-device / pipelined
// overlap; the caller must hipStreamSynchronize that stream before reading
// winsums_host). slot selects the per-device pipeline slot (scratch +
// stream); slot 1's stream is created lazily.
int msm_batch_shard_device(spectre_gpu_ctx* ctx, int dev,
                           const g1_affine* d_bases, const uint8_t* d_scalars,
                           uint32_t nbatch, uint64_t n, uint32_t flags,
                           g1_jac* winsums_host, double* stage_ms = nullptr,
                           bool sync = true, int slot = 0);

// msm.hip — window-ranged pipeline: only windows [w_lo, w_lo+w_cnt) are
// decomposed/accumulated (window-sharded multi-GPU: bucket work AND the
// reduction tail divide by the shard count; the carry recoding still runs
// over all windows). winsums_host receives nbatch * w_cnt Jacobian sums.
int msm_batch_windows_device(spectre_gpu_ctx* ctx, int dev,
                             const g1_affine* d_bases,
                             const uint8_t* d_scalars, uint32_t nbatch,
                             uint64_t n, uint32_t flags, uint32_t w_lo,
                             uint32_t w_cnt, g1_jac* winsums_host,
                             double* stage_ms = nullptr, bool sync = true,
                             int slot = 0);

// msm.hip — sync a slot's stream and deliver pending window sums.
int msm_slot_drain(spectre_gpu_ctx* ctx, int dev, int slot);

// ntt.hip — gate-expression evaluator over device column buffers
// (synchronizes). cols = host array of ncols device pointers.
int fr_gate_eval_device(spectre_gpu_ctx* ctx, int dev,
                        const fp256* const* cols, uint32_t ncols,
                        const fp256* consts, uint32_t nconst,
                        const uint32_t* program, uint32_t nops, uint64_t n,
                        uint32_t rot_scale, const fp256* y, fp256* d_out);

// ntt.hip — pointwise Fr vector op on device buffers (synchronizes).
int fr_vec_op_device(spectre_gpu_ctx* ctx, int dev, int op, const fp256* d_a,
                     const fp256* d_b, const fp256* c, fp256* d_out,
                     uint64_t n);
// ntt.hip — in-place NTT on a device buffer (synchronizes the stream).
int ntt_device(spectre_gpu_ctx* ctx, int dev, fp256* d_data, uint32_t log_n,
               const fp256& omega, int inverse, const fp256* coset_gen);
