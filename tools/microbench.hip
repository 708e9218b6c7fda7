// tools/microbench.hip — standalone probe: how many Jacobian+affine mixed
// adds per second can one gfx950 sustain, as a function of independent
// add-chains per thread (ILP)? Informs the bucket-accumulation design.
// Build+run (GPU box):
//   hipcc -O3 --offload-arch=gfx950 -I include -I spectre_amd/csrc \
//       tools/microbench.hip -o gpurun_out/microbench && gpurun_out/microbench
#include <hip/hip_runtime.h>
#include <cstdio>
#include "../spectre_amd/csrc/g1.hpp"

#define ITERS 256


// raw v_mad_u64_u32 issue-rate probe: 4 independent mad chains per thread
__global__ __launch_bounds__(256, 4) void k_mad64_rate(uint64_t* out,
                                                       uint32_t iters) {
    uint64_t a0 = threadIdx.x | 1, a1 = a0 + 3, a2 = a0 + 5, a3 = a0 + 7;
    const uint32_t b = blockIdx.x | 3;
    for (uint32_t i = 0; i < iters; i++) {
        a0 = (uint64_t)(uint32_t)a0 * b + a1;
        a1 = (uint64_t)(uint32_t)a1 * b + a2;
        a2 = (uint64_t)(uint32_t)a2 * b + a3;
        a3 = (uint64_t)(uint32_t)a3 * b + a0;
    }
    out[blockIdx.x * blockDim.x + threadIdx.x] = a0 + a1 + a2 + a3;
}

template <int CHAINS>
__global__ __launch_bounds__(256, 2) void k_madd_chain(const g1_affine* pts,
                                                       g1_jac* out, int nwork) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= nwork) return;
    g1_affine q[CHAINS];
    g1_jac acc[CHAINS];
    for (int c = 0; c < CHAINS; c++) {
        q[c] = pts[(t + c * 37) % 1024];
        g1j_from_affine(acc[c], q[c]);
    }
    for (int i = 0; i < ITERS; i++) {
#pragma unroll
        for (int c = 0; c < CHAINS; c++) g1j_madd_ip(acc[c], q[c]);
    }
    for (int c = 0; c < CHAINS; c++) out[t * CHAINS + c] = acc[c];
}

// pure CIOS multiply chain (field-mul latency floor)
__global__ __launch_bounds__(256, 2) void k_mul_chain(const g1_affine* pts,
                                                      g1_jac* out, int nwork) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= nwork) return;
    fp256 a = pts[t % 1024].x, b = pts[t % 1024].y;
    for (int i = 0; i < ITERS * 16; i++) ff_mul_cios<Fq>(a, a, b);
    out[t].X = a;
}

// product ff_mul (asm column-Montgomery on device) chain
__global__ __launch_bounds__(256, 2) void k_mul_cols_chain(
    const g1_affine* pts, g1_jac* out, int nwork) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= nwork) return;
    fp256 a = pts[t % 1024].x, b = pts[t % 1024].y;
    for (int i = 0; i < ITERS * 16; i++) ff_mul<Fq>(a, a, b);  // asm path
    out[t].X = a;
}

// two INDEPENDENT product-path mul chains per thread: if this runs much
// faster than 2x the single-chain time, the single chain is stall-bound
// (dependent-mad latency) and a dual-accumulator column schedule would pay;
// if it matches, the mul is issue-bound and deeper ILP buys nothing.
__global__ __launch_bounds__(256, 2) void k_mul2_chain(const g1_affine* pts,
                                                       g1_jac* out, int nwork) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= nwork) return;
    fp256 a = pts[t % 1024].x, b = pts[t % 1024].y;
    fp256 a2 = pts[(t + 7) % 1024].y, b2 = pts[(t + 7) % 1024].x;
    for (int i = 0; i < ITERS * 8; i++) {
        ff_mul<Fq>(a, a, b);
        ff_mul<Fq>(a2, a2, b2);
    }
    out[t].X = a;
    out[t].Y = a2;
}

// correctness: ff_mul_cols must equal ff_mul bit-for-bit on pseudorandom
// reduced inputs (both Fq and Fr), chained so errors compound and surface.
__global__ void k_mul_cols_check(uint32_t* bad, int nwork) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= nwork) return;
    uint64_t s = 0x9e3779b97f4a7c15ull * (t + 1);
    fp256 a, b;
    for (int i = 0; i < 8; i++) {
        s = s * 6364136223846793005ull + 1442695040888963407ull;
        a.l[i] = (uint32_t)(s >> 32);
        s = s * 6364136223846793005ull + 1442695040888963407ull;
        b.l[i] = (uint32_t)(s >> 32);
    }
    a.l[7] &= 0x0fffffffu;  // < 2^252 < p (both moduli)
    b.l[7] &= 0x0fffffffu;
    fp256 xq = a, yq = a, xr = b, yr = b;
    for (int i = 0; i < 64; i++) {
        ff_mul_cios<Fq>(xq, xq, b);
        ff_mul<Fq>(yq, yq, b);  // asm column form (product path)
        ff_mul_cios<Fr>(xr, xr, a);
        ff_mul<Fr>(yr, yr, a);
    }
    if (!ff_eq(xq, yq) || !ff_eq(xr, yr)) atomicAdd(bad, 1u);
}

static double time_kernel(void (*fn)(const g1_affine*, g1_jac*, int),
                          const g1_affine* pts, g1_jac* out, int nwork,
                          int blocks) {
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    hipLaunchKernelGGL(fn, dim3(blocks), dim3(256), 0, 0, pts, out, nwork);
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(fn, dim3(blocks), dim3(256), 0, 0, pts, out, nwork);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms;
    (void)hipEventElapsedTime(&ms, e0, e1);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return ms;
}

int main() {
    // fabricate valid-shaped data (values need not be on-curve for a
    // throughput probe; formulas don't branch on curve membership)
    g1_affine* pts;
    g1_jac* out;
    (void)hipMalloc(&pts, 1024 * sizeof(g1_affine));
    (void)hipMalloc(&out, (size_t)4 * 1024 * 1024 * sizeof(g1_jac));
    (void)hipMemset(pts, 0x5a, 1024 * sizeof(g1_affine));
    for (int nwork : {256 * 1024, 1024 * 1024}) {
        int blocks = nwork / 256;
        double ms1 = time_kernel(k_madd_chain<1>, pts, out, nwork, blocks);
        double ms2 = time_kernel(k_madd_chain<2>, pts, out, nwork / 2, blocks / 2);
        double ms4 = time_kernel(k_madd_chain<4>, pts, out, nwork / 4, blocks / 4);
        double madds = (double)nwork * ITERS;
        printf("nwork=%7d  1-chain %.3f ms (%.1f M madd/s)  2-chain %.3f ms "
               "(%.1f M madd/s)  4-chain %.3f ms (%.1f M madd/s)\n",
               nwork, ms1, madds / ms1 / 1e3, ms2, madds / ms2 / 1e3, ms4,
               madds / ms4 / 1e3);
    }
    int nwork = 1024 * 1024;
    double msm = time_kernel(k_mul_chain, pts, out, nwork, nwork / 256);
    printf("C CIOS chain: %.3f ms  (%.1f M mul/s)\n", msm,
           (double)nwork * ITERS * 16 / msm / 1e3);
    {
        uint32_t* bad;
        (void)hipMalloc(&bad, 4);
        (void)hipMemset(bad, 0, 4);
        hipLaunchKernelGGL(k_mul_cols_check, dim3(1024), dim3(256), 0, 0, bad,
                           1024 * 256);
        uint32_t h_bad = 1;
        (void)hipMemcpy(&h_bad, bad, 4, hipMemcpyDeviceToHost);
        printf("asm ff_mul parity vs C CIOS (256K lanes x 64 chained, "
               "Fq+Fr): %s (%u bad)\n", h_bad ? "FAIL" : "ok", h_bad);
        double msc = time_kernel(k_mul_cols_chain, pts, out, nwork, nwork / 256);
        printf("asm ff_mul chain: %.3f ms  (%.1f M mul/s, %.2fx vs C)\n",
               msc, (double)nwork * ITERS * 16 / msc / 1e3, msm / msc);
        double ms2 = time_kernel(k_mul2_chain, pts, out, nwork, nwork / 256);
        printf("asm ff_mul 2-indep chains: %.3f ms  (%.1f M mul/s total; "
               "stall-bound if >> the 1-chain rate)\n",
               ms2, (double)nwork * ITERS * 16 / ms2 / 1e3);
        (void)hipFree(bad);
    }
    // raw mad64 rate
    {
        hipEvent_t e0, e1;
        (void)hipEventCreate(&e0);
        (void)hipEventCreate(&e1);
        uint32_t iters = 4096;
        hipLaunchKernelGGL(k_mad64_rate, dim3(nwork / 256), dim3(256), 0, 0,
                           (uint64_t*)out, iters);
        (void)hipDeviceSynchronize();
        (void)hipEventRecord(e0);
        hipLaunchKernelGGL(k_mad64_rate, dim3(nwork / 256), dim3(256), 0, 0,
                           (uint64_t*)out, iters);
        (void)hipEventRecord(e1);
        (void)hipEventSynchronize(e1);
        float ms;
        (void)hipEventElapsedTime(&ms, e0, e1);
        double mads = (double)nwork * iters * 4;
        printf("v_mad_u64_u32 (4 indep chains): %.3f ms  %.2f T mad64/s "
               "(peak-if-2cyc = 78.6T, 4cyc = 39.3T, 8cyc = 19.7T)\n", ms,
               mads / ms / 1e9);
    }
    return 0;
}
