// ffi.cpp — C-ABI implementation of libspectre_gpu.so (see
// include/spectre_gpu.h for the boundary contract and INTEGRATION.md for the
// Rust-side binding that slots this beneath halo2's best_multiexp/best_fft).
//
// Host-side work is deliberately tiny: context/stream/cache management,
// SRS upload caching, and the final window-Horner (<= 256 point ops per MSM)
// + one field inversion to normalize to affine. All bulk compute runs in the
// HIP kernels (msm.hip / ntt.hip); there is no CPU fallback path.
#include "internal.hpp"
#include "spectre_gpu.h"
#include <cstdarg>
#include <cstdlib>
#include <cstdio>

thread_local std::string g_last_error;
void set_err(const char* fmt, ...) {
    char buf[512];
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(buf, sizeof buf, fmt, ap);
    va_end(ap);
    g_last_error = buf;
}

extern "C" {

const char* spectre_gpu_last_error(void) { return g_last_error.c_str(); }
const char* spectre_gpu_version(void) { return "spectre-amd 0.1.0 gfx950"; }

spectre_gpu_ctx* spectre_gpu_init(int device_count, const int* device_ids) {
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev == 0) {
        set_err("spectre_gpu_init: no visible HIP devices (this library has "
                "no CPU fallback by design)");
        return nullptr;
    }
    if (device_count <= 0) device_count = 1;
    auto* ctx = new spectre_gpu_ctx();
    for (int i = 0; i < device_count; i++) {
        int id = device_ids ? device_ids[i] : i;
        if (id < 0 || id >= ndev) {
            set_err("spectre_gpu_init: device id %d out of range (%d devices)",
                    id, ndev);
            delete ctx;
            return nullptr;
        }
        DeviceState ds;
        ds.device_id = id;
        if (hipSetDevice(id) != hipSuccess ||
            hipStreamCreate(&ds.stream) != hipSuccess) {
            set_err("spectre_gpu_init: failed to init device %d", id);
            delete ctx;
            return nullptr;
        }
        ctx->devs.push_back(ds);
    }
    return ctx;
}

void spectre_gpu_destroy(spectre_gpu_ctx* ctx) {
    if (!ctx) return;
    for (auto& ds : ctx->devs) {
        (void)hipSetDevice(ds.device_id);
        for (auto& sl : ds.slots) {
            for (void* p : {(void*)sl.d_keys_in, (void*)sl.d_keys_out,
                            (void*)sl.d_vals_in, (void*)sl.d_vals_out,
                            (void*)sl.d_sort_tmp, (void*)sl.d_offsets,
                            (void*)sl.d_buckets, (void*)sl.d_firstK,
                            (void*)sl.d_lastK, (void*)sl.d_firstP,
                            (void*)sl.d_lastP, (void*)sl.d_red})
                if (p) (void)hipFree(p);
            if (sl.h_wins) (void)hipHostFree(sl.h_wins);
            if (sl.stream && sl.stream != ds.stream)
                (void)hipStreamDestroy(sl.stream);
        }
        for (void* p : {(void*)ds.d_scalars, (void*)ds.d_bases,
                        (void*)ds.d_ntt_tmp, (void*)ds.d_ntt_io,
                        (void*)ds.d_cosetA})
            if (p) (void)hipFree(p);
        for (int i = 0; i < 2; i++) {
            if (ds.h_stage[i]) (void)hipHostFree(ds.h_stage[i]);
            if (ds.stage_ev[i]) (void)hipEventDestroy(ds.stage_ev[i]);
        }
        for (auto& kv : ds.bases_cache)
            if (kv.second.d_ptr) (void)hipFree(kv.second.d_ptr);
        for (auto& kv : ds.plans) {
            if (kv.second.tw1) (void)hipFree(kv.second.tw1);
            if (kv.second.tw2) (void)hipFree(kv.second.tw2);
            if (kv.second.tw3) (void)hipFree(kv.second.tw3);
            if (kv.second.twB) (void)hipFree(kv.second.twB);
        }
        if (ds.stream) (void)hipStreamDestroy(ds.stream);
    }
    delete ctx;
}

int spectre_gpu_device_count(spectre_gpu_ctx* ctx) {
    return ctx ? (int)ctx->devs.size() : 0;
}

// ---------------------------------------------------------------- staging
#define STAGE_CHUNK (16u << 20)

static int ensure_stage(DeviceState& ds) {
    if (!ds.h_stage[0]) {
        HIP_TRY(hipHostMalloc((void**)&ds.h_stage[0], STAGE_CHUNK));
        HIP_TRY(hipHostMalloc((void**)&ds.h_stage[1], STAGE_CHUNK));
        HIP_TRY(hipEventCreate(&ds.stage_ev[0]));
        HIP_TRY(hipEventCreate(&ds.stage_ev[1]));
        HIP_TRY(hipEventRecord(ds.stage_ev[0], ds.stream));
        HIP_TRY(hipEventRecord(ds.stage_ev[1], ds.stream));
    }
    return 0;
}

// host->device through alternating pinned chunks: memcpy of chunk k+1
// overlaps the DMA of chunk k; the stream stays ordered so following
// kernels need no extra sync. The user buffer is fully consumed by the
// memcpys before return (the in-flight DMA reads only pinned memory).
// double-buffered pinned-staging copies on ds.stream: pageable hipMemcpy
// runs at ~3 GB/s; staging through pinned chunks overlaps the host memcpy
// with the DMA and reaches ~3-4x that.
static int staged_upload(DeviceState& ds, void* d_dst, const void* src,
                         size_t bytes) {
    int rc = ensure_stage(ds);
    if (rc) return rc;
    size_t off = 0;
    int b = 0;
    while (off < bytes) {
        const size_t len = bytes - off < STAGE_CHUNK ? bytes - off : STAGE_CHUNK;
        HIP_TRY(hipEventSynchronize(ds.stage_ev[b]));  // buffer free?
        memcpy(ds.h_stage[b], (const uint8_t*)src + off, len);
        HIP_TRY(hipMemcpyAsync((uint8_t*)d_dst + off, ds.h_stage[b], len,
                               hipMemcpyHostToDevice, ds.stream));
        HIP_TRY(hipEventRecord(ds.stage_ev[b], ds.stream));
        off += len;
        b ^= 1;
    }
    return 0;
}

static int staged_download(DeviceState& ds, void* dst, const void* d_src,
                           size_t bytes) {
    int rc = ensure_stage(ds);
    if (rc) return rc;
    size_t off = 0;
    int b = 0;
    size_t pend_off[2] = {0, 0}, pend_len[2] = {0, 0};
    while (off < bytes) {
        const size_t len = bytes - off < STAGE_CHUNK ? bytes - off : STAGE_CHUNK;
        if (pend_len[b]) {  // drain the older use of this buffer
            HIP_TRY(hipEventSynchronize(ds.stage_ev[b]));
            memcpy((uint8_t*)dst + pend_off[b], ds.h_stage[b], pend_len[b]);
        }
        HIP_TRY(hipMemcpyAsync(ds.h_stage[b], (const uint8_t*)d_src + off, len,
                               hipMemcpyDeviceToHost, ds.stream));
        HIP_TRY(hipEventRecord(ds.stage_ev[b], ds.stream));
        pend_off[b] = off;
        pend_len[b] = len;
        off += len;
        b ^= 1;
    }
    for (int i = 0; i < 2; i++) {
        if (pend_len[i]) {
            HIP_TRY(hipEventSynchronize(ds.stage_ev[i]));
            memcpy((uint8_t*)dst + pend_off[i], ds.h_stage[i], pend_len[i]);
        }
    }
    return 0;
}

// ---------------------------------------------------------------- helpers
static int check_dev(spectre_gpu_ctx* ctx, int dev) {
    if (!ctx || dev < 0 || dev >= (int)ctx->devs.size()) {
        set_err("invalid ctx/device index %d", dev);
        return -1;
    }
    return 0;
}

// Horner over the 16 windows + normalize to affine (host; <=256 point ops).
static void winsums_to_affine(const g1_jac* wins, uint8_t out[64]) {
    g1_jac res;
    g1j_set_inf(res);
    for (int w = MSM_NWIN - 1; w >= 0; w--) {
        if (w != MSM_NWIN - 1) {
            for (int d = 0; d < MSM_WBITS; d++) {
                g1_jac t = res;
                g1j_dbl(res, t);
            }
        }
        g1_jac t = res;
        g1j_add(res, t, wins[w]);
    }
    g1_affine a;
    g1j_to_affine(a, res);
    ff_to_bytes(out, a.x);
    ff_to_bytes(out + 32, a.y);
}

int spectre_gpu_msm_g1_combine(const uint8_t* partials, uint32_t nshards,
                               uint8_t out_affine[64]) {
    if (!partials || nshards == 0) {
        set_err("combine: bad args");
        return -1;
    }
    g1_jac tot[MSM_NWIN];
    for (int w = 0; w < MSM_NWIN; w++) g1j_set_inf(tot[w]);
    for (uint32_t s = 0; s < nshards; s++) {  // deterministic rank order
        const g1_jac* sh = (const g1_jac*)(partials + (size_t)s * MSM_NWIN * 96);
        for (int w = 0; w < MSM_NWIN; w++) {
            g1_jac t = tot[w];
            g1j_add(tot[w], t, sh[w]);
        }
    }
    winsums_to_affine(tot, out_affine);
    return 0;
}

// ---------------------------------------------------------------- MSM
int spectre_gpu_msm_g1_shard_device(spectre_gpu_ctx* ctx, int dev,
                                    const void* d_bases, const void* d_scalars,
                                    uint64_t n, uint32_t flags,
                                    uint8_t* out_partials) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    return msm_shard_device(ctx, dev, (const g1_affine*)d_bases,
                            (const uint8_t*)d_scalars, n, flags,
                            (g1_jac*)out_partials);
}

int spectre_gpu_msm_g1_shard_device_timed(spectre_gpu_ctx* ctx, int dev,
                                          const void* d_bases,
                                          const void* d_scalars, uint64_t n,
                                          uint32_t flags,
                                          uint8_t* out_partials,
                                          double out_ms[8]) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    return msm_shard_device(ctx, dev, (const g1_affine*)d_bases,
                            (const uint8_t*)d_scalars, n, flags,
                            (g1_jac*)out_partials, out_ms);
}

int spectre_gpu_msm_g1_shard_device_async(spectre_gpu_ctx* ctx, int dev,
                                          const void* d_bases,
                                          const void* d_scalars, uint64_t n,
                                          uint32_t flags,
                                          uint8_t* out_partials,
                                          int* out_slot) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    DeviceState& ds = ctx->devs[dev];
    // measured on MI355X (r2): depth 3 = 325.6 MSM(2^20)/s vs depth 2 =
    // 300 vs unpipelined = 240 — three in flight keep the machine fed
    // through each call's host-combine + sync gap.
    static const int kSlots = []() {
        const char* e = getenv("SPECTRE_PIPE_SLOTS");
        int v = e ? atoi(e) : 3;
        return v < 1 ? 1 : (v > 3 ? 3 : v);
    }();
    const int slot = ds.next_slot;
    ds.next_slot = (ds.next_slot + 1) % kSlots;
    *out_slot = slot;
    return msm_batch_shard_device(ctx, dev, (const g1_affine*)d_bases,
                                  (const uint8_t*)d_scalars, 1, n, flags,
                                  (g1_jac*)out_partials, nullptr,
                                  /*sync=*/false, slot);
}

// window-sharded multi-GPU: rank r of R computes windows
// [r*NWIN/R, (r+1)*NWIN/R) over ALL n points. out_partials = w_cnt * 96 B.
int spectre_gpu_msm_g1_shard_windows_device(spectre_gpu_ctx* ctx, int dev,
                                            const void* d_bases,
                                            const void* d_scalars, uint64_t n,
                                            uint32_t flags, uint32_t w_lo,
                                            uint32_t w_cnt,
                                            uint8_t* out_partials) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    return msm_batch_windows_device(ctx, dev, (const g1_affine*)d_bases,
                                    (const uint8_t*)d_scalars, 1, n, flags,
                                    w_lo, w_cnt, (g1_jac*)out_partials);
}

int spectre_gpu_msm_g1_shard_windows_device_async(
    spectre_gpu_ctx* ctx, int dev, const void* d_bases, const void* d_scalars,
    uint64_t n, uint32_t flags, uint32_t w_lo, uint32_t w_cnt,
    uint8_t* out_partials, int* out_slot) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    DeviceState& ds = ctx->devs[dev];
    static const int kSlots = []() {
        const char* e = getenv("SPECTRE_PIPE_SLOTS");
        int v = e ? atoi(e) : 3;
        return v < 1 ? 1 : (v > 3 ? 3 : v);
    }();
    const int slot = ds.next_slot;
    ds.next_slot = (ds.next_slot + 1) % kSlots;
    *out_slot = slot;
    return msm_batch_windows_device(ctx, dev, (const g1_affine*)d_bases,
                                    (const uint8_t*)d_scalars, 1, n, flags,
                                    w_lo, w_cnt, (g1_jac*)out_partials,
                                    nullptr, /*sync=*/false, slot);
}

// assemble disjoint per-shard window slices (shard i = NWIN/nshards
// consecutive windows, rank order) and finish with the window Horner.
int spectre_gpu_msm_g1_combine_windows(const uint8_t* partials,
                                       uint32_t nshards,
                                       uint8_t out_affine[64]) {
    if (!partials || nshards == 0 || MSM_NWIN % nshards != 0) {
        set_err("combine_windows: nshards must divide %d", MSM_NWIN);
        return -1;
    }
    winsums_to_affine((const g1_jac*)partials, out_affine);
    return 0;
}

int spectre_gpu_msm_slot_wait(spectre_gpu_ctx* ctx, int dev, int slot) {
    if (check_dev(ctx, dev)) return -1;
    if (slot < 0 || slot > 2) {
        set_err("slot_wait: slot %d out of range", slot);
        return -1;
    }
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    return msm_slot_drain(ctx, dev, slot);
}

int spectre_gpu_msm_g1_device(spectre_gpu_ctx* ctx, int dev,
                              const void* d_bases, const void* d_scalars,
                              uint64_t n, uint32_t flags,
                              uint8_t out_affine[64]) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    g1_jac wins[MSM_NWIN];
    int rc = msm_shard_device(ctx, dev, (const g1_affine*)d_bases,
                              (const uint8_t*)d_scalars, n, flags, wins);
    if (rc) return rc;
    winsums_to_affine(wins, out_affine);
    return 0;
}

int spectre_gpu_msm_g1_batch_device(spectre_gpu_ctx* ctx, int dev,
                                    const void* d_bases,
                                    const void* d_scalars, uint32_t nbatch,
                                    uint64_t n, uint32_t flags,
                                    uint8_t* out_affine) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    std::vector<g1_jac> wins((size_t)nbatch * MSM_NWIN);
    int rc = msm_batch_shard_device(ctx, dev, (const g1_affine*)d_bases,
                                    (const uint8_t*)d_scalars, nbatch, n,
                                    flags, wins.data());
    if (rc) return rc;
    for (uint32_t b = 0; b < nbatch; b++)
        winsums_to_affine(&wins[(size_t)b * MSM_NWIN], out_affine + 64 * b);
    return 0;
}

int spectre_gpu_msm_g1_batch(spectre_gpu_ctx* ctx, uint64_t bases_id,
                             const uint8_t* bases, const uint8_t* scalars,
                             uint32_t nbatch, uint64_t n, uint32_t flags,
                             uint8_t* out_affine) {
    if (!ctx) {
        set_err("null ctx");
        return -1;
    }
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    if (n == 0) {
        memset(out_affine, 0, (size_t)64 * nbatch);
        return 0;
    }
    if (!scalars) {
        set_err("scalars is NULL");
        return -1;
    }
    DeviceState& ds = ctx->devs[0];
    HIP_TRY(hipSetDevice(ds.device_id));
    const uint64_t sbytes = (uint64_t)nbatch * n * 32;
    if (ds.scal_cap < sbytes) {
        if (ds.d_scalars) (void)hipFree(ds.d_scalars);
        HIP_TRY(hipMalloc(&ds.d_scalars, sbytes));
        ds.scal_cap = sbytes;
    }
    {
        int rc = staged_upload(ds, ds.d_scalars, scalars, sbytes);
        if (rc) return rc;
    }
    g1_affine* d_b = nullptr;
    if (bases_id != 0) {
        std::array<uint64_t, 3> key{bases_id, n, 1};
        auto it = ds.bases_cache.find(key);
        if (it != ds.bases_cache.end()) {
            d_b = it->second.d_ptr;
        } else {
            if (!bases) {
                set_err("bases_id %llu not cached and bases is NULL",
                        (unsigned long long)bases_id);
                return -4;
            }
            CachedBases cb;
            cb.n = n;
            HIP_TRY(hipMalloc(&cb.d_ptr, n * sizeof(g1_affine)));
            int rc = staged_upload(ds, cb.d_ptr, bases, n * 64);
            if (rc) return rc;
            ds.bases_cache.emplace(key, cb);
            d_b = cb.d_ptr;
        }
    } else {
        if (!bases) {
            set_err("bases is NULL");
            return -1;
        }
        if (ds.base_cap < n) {
            if (ds.d_bases) (void)hipFree(ds.d_bases);
            HIP_TRY(hipMalloc(&ds.d_bases, n * sizeof(g1_affine)));
            ds.base_cap = n;
        }
        int rc = staged_upload(ds, ds.d_bases, bases, n * 64);
        if (rc) return rc;
        d_b = ds.d_bases;
    }
    return spectre_gpu_msm_g1_batch_device(ctx, 0, d_b, ds.d_scalars, nbatch,
                                           n, flags, out_affine);
}

// upload + enqueue one device's shard of an n-point MSM split num_gpus ways
// (returns with async work in flight on that device's stream).
static int msm_multi_enqueue_one(spectre_gpu_ctx* ctx, int d, int num_gpus,
                                 uint64_t bases_id, const uint8_t* bases,
                                 const uint8_t* scalars, uint64_t n,
                                 uint32_t flags, g1_jac* wins_out) {
    DeviceState& ds = ctx->devs[d];
    HIP_TRY(hipSetDevice(ds.device_id));
    const uint64_t lo = n * d / num_gpus, hi = n * (d + 1) / num_gpus;
    const uint64_t m = hi - lo;
    // scalars: plain scratch upload
    if (ds.scal_cap < m * 32) {
        if (ds.d_scalars) { (void)hipFree(ds.d_scalars); ds.d_scalars = nullptr; }
        ds.scal_cap = 0;
        HIP_TRY(hipMalloc(&ds.d_scalars, m * 32));
        ds.scal_cap = m * 32;
    }
    {
        int rc = staged_upload(ds, ds.d_scalars, scalars + lo * 32, m * 32);
        if (rc) return rc;
    }
    // bases: cached per (bases_id, n, num_gpus) — this device's chunk layout
    // depends on all three, so all three form the cache key.
    g1_affine* d_b = nullptr;
    if (bases_id != 0) {
        std::array<uint64_t, 3> key{bases_id, n, (uint64_t)num_gpus};
        auto it = ds.bases_cache.find(key);
        if (it != ds.bases_cache.end()) {
            d_b = it->second.d_ptr;
        } else {
            if (!bases) {
                set_err("bases_id %llu not cached and bases is NULL",
                        (unsigned long long)bases_id);
                return -4;
            }
            CachedBases cb;
            cb.n = m;
            HIP_TRY(hipMalloc(&cb.d_ptr, m * sizeof(g1_affine)));
            int rc = staged_upload(ds, cb.d_ptr, bases + lo * 64, m * 64);
            if (rc) return rc;
            ds.bases_cache.emplace(key, cb);
            d_b = cb.d_ptr;
        }
    } else {
        if (!bases) {
            set_err("bases is NULL");
            return -1;
        }
        if (ds.base_cap < m) {
            if (ds.d_bases) { (void)hipFree(ds.d_bases); ds.d_bases = nullptr; }
            ds.base_cap = 0;
            HIP_TRY(hipMalloc(&ds.d_bases, m * sizeof(g1_affine)));
            ds.base_cap = m;
        }
        int rc = staged_upload(ds, ds.d_bases, bases + lo * 64, m * 64);
        if (rc) return rc;
        d_b = ds.d_bases;
    }
    // enqueue without synchronizing so all shards run concurrently
    return msm_batch_shard_device(ctx, d, d_b, ds.d_scalars, 1, m, flags,
                                  wins_out, nullptr, /*sync=*/false);
}

int spectre_gpu_msm_g1(spectre_gpu_ctx* ctx, uint64_t bases_id,
                       const uint8_t* bases, const uint8_t* scalars, uint64_t n,
                       uint32_t flags, int num_gpus, uint8_t out_affine[64]) {
    if (!ctx) {
        set_err("null ctx");
        return -1;
    }
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    if (n == 0) {
        memset(out_affine, 0, 64);
        return 0;
    }
    if (num_gpus <= 0) num_gpus = 1;
    if (num_gpus > (int)ctx->devs.size()) {
        set_err("num_gpus %d > ctx devices %zu", num_gpus, ctx->devs.size());
        return -1;
    }
    if (!scalars) {
        set_err("scalars is NULL");
        return -1;
    }
    std::vector<g1_jac> wins((size_t)num_gpus * MSM_NWIN);
    for (int d = 0; d < num_gpus; d++) {
        int rc = msm_multi_enqueue_one(ctx, d, num_gpus, bases_id, bases,
                                       scalars, n, flags,
                                       &wins[(size_t)d * MSM_NWIN]);
        if (rc) {
            // Devices <= d may have kernels and an async D2H into `wins`
            // in flight; drain them before the vector goes out of scope.
            for (int e = 0; e <= d; e++) {
                (void)hipSetDevice(ctx->devs[e].device_id);
                (void)hipStreamSynchronize(ctx->devs[e].stream);
            }
            return rc;
        }
    }
    for (int d = 0; d < num_gpus; d++) {
        int rc = msm_slot_drain(ctx, d, 0);
        if (rc) return rc;
    }
    return spectre_gpu_msm_g1_combine((const uint8_t*)wins.data(),
                                      (uint32_t)num_gpus, out_affine);
}

// ---------------------------------------------------------------- NTT
int spectre_gpu_ntt_fr_device(spectre_gpu_ctx* ctx, int dev, void* d_data,
                              uint32_t log_n, const uint8_t omega[32],
                              int inverse, const uint8_t* coset_gen) {
    if (check_dev(ctx, dev)) return -1;
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    fp256 om, g;
    ff_from_bytes(om, omega);
    if (coset_gen) ff_from_bytes(g, coset_gen);
    return ntt_device(ctx, dev, (fp256*)d_data, log_n, om, inverse,
                      coset_gen ? &g : nullptr);
}

int spectre_gpu_ntt_fr(spectre_gpu_ctx* ctx, uint8_t* data, uint32_t log_n,
                       const uint8_t omega[32], int inverse,
                       const uint8_t* coset_gen) {
    if (check_dev(ctx, 0)) return -1;
    if (log_n > 28) {
        set_err("ntt: log_n %u > 28 unsupported (Fr 2-adicity)", log_n);
        return -3;
    }
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    DeviceState& ds = ctx->devs[0];
    HIP_TRY(hipSetDevice(ds.device_id));
    const uint64_t n = 1ull << log_n;
    // cached IO buffer: the Rust seam calls this entry ~40-60x per proof and
    // must not pay a (up to 1 GB) hipMalloc/hipFree per transform.
    if (ds.ntt_io_cap < n) {
        if (ds.d_ntt_io) { (void)hipFree(ds.d_ntt_io); ds.d_ntt_io = nullptr; }
        ds.ntt_io_cap = 0;
        HIP_TRY(hipMalloc(&ds.d_ntt_io, n * 32));
        ds.ntt_io_cap = n;
    }
    int rc = staged_upload(ds, ds.d_ntt_io, data, n * 32);
    if (rc == 0)
        rc = spectre_gpu_ntt_fr_device(ctx, 0, ds.d_ntt_io, log_n, omega,
                                       inverse, coset_gen);
    if (rc == 0) rc = staged_download(ds, data, ds.d_ntt_io, n * 32);
    return rc;
}

int spectre_gpu_fr_gate_eval(spectre_gpu_ctx* ctx, int dev,
                             const void* const* d_cols, uint32_t ncols,
                             const uint8_t* constants, uint32_t nconst,
                             const uint32_t* program, uint32_t nops,
                             uint64_t n, uint32_t rot_scale, const uint8_t* y,
                             void* d_out) {
    if (check_dev(ctx, dev)) return -1;
    if (!d_cols || !program || !nops || !d_out || (nconst && !constants)) {
        set_err("gate_eval: bad arguments");
        return -1;
    }
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    fp256 yv;
    if (y) ff_from_bytes(yv, y);
    return fr_gate_eval_device(ctx, dev, (const fp256* const*)d_cols, ncols,
                               (const fp256*)constants, nconst, program, nops,
                               n, rot_scale, y ? &yv : nullptr,
                               (fp256*)d_out);
}

int spectre_gpu_fr_vec_op(spectre_gpu_ctx* ctx, int dev, int op,
                          const void* d_a, const void* d_b, const uint8_t* c,
                          void* d_out, uint64_t n) {
    if (check_dev(ctx, dev)) return -1;
    if (op < 0 || op > 4 || !d_a || !d_out ||
        (op <= 2 && !d_b) || (op == 4 && !d_b) || (op >= 3 && !c)) {
        set_err("fr_vec_op: bad op/arguments");
        return -1;
    }
    std::lock_guard<std::recursive_mutex> lk(ctx->mu);
    fp256 cv;
    if (c) ff_from_bytes(cv, c);
    return fr_vec_op_device(ctx, dev, op, (const fp256*)d_a,
                            (const fp256*)d_b, c ? &cv : nullptr,
                            (fp256*)d_out, n);
}

// ---------------------------------------------------------------- memory
int spectre_gpu_malloc(spectre_gpu_ctx* ctx, int dev, size_t bytes,
                       void** d_ptr) {
    if (check_dev(ctx, dev)) return -1;
    HIP_TRY(hipSetDevice(ctx->devs[dev].device_id));
    HIP_TRY(hipMalloc(d_ptr, bytes));
    return 0;
}
int spectre_gpu_free(spectre_gpu_ctx* ctx, int dev, void* d_ptr) {
    if (check_dev(ctx, dev)) return -1;
    HIP_TRY(hipSetDevice(ctx->devs[dev].device_id));
    HIP_TRY(hipFree(d_ptr));
    return 0;
}
int spectre_gpu_upload(spectre_gpu_ctx* ctx, int dev, void* d_dst,
                       const void* src, size_t bytes) {
    if (check_dev(ctx, dev)) return -1;
    HIP_TRY(hipSetDevice(ctx->devs[dev].device_id));
    HIP_TRY(hipMemcpy(d_dst, src, bytes, hipMemcpyHostToDevice));
    return 0;
}
int spectre_gpu_download(spectre_gpu_ctx* ctx, int dev, void* dst,
                         const void* d_src, size_t bytes) {
    if (check_dev(ctx, dev)) return -1;
    HIP_TRY(hipSetDevice(ctx->devs[dev].device_id));
    HIP_TRY(hipMemcpy(dst, d_src, bytes, hipMemcpyDeviceToHost));
    return 0;
}
int spectre_gpu_synchronize(spectre_gpu_ctx* ctx, int dev) {
    if (check_dev(ctx, dev)) return -1;
    HIP_TRY(hipSetDevice(ctx->devs[dev].device_id));
    HIP_TRY(hipDeviceSynchronize());
    return 0;
}

}  // extern "C"
