// mfma_peak — fp64 MFMA dense-throughput microbenchmark for gfx950.
// Measures the achievable v_mfma_f64_16x16x4_f64 rate (the real roofline
// ceiling under DVFS, vs the 78.6 TF/s spec at 2.4 GHz).
//   hipcc --offload-arch=gfx950 -O3 mfma_peak.hip -o mfma_peak && ./mfma_peak
#include <hip/hip_runtime.h>

#include <cstdio>

typedef double f64x4 __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(256) void peak_kernel(const double *in,
                                                   double *out, int iters) {
    const double a = in[threadIdx.x & 63];
    const double b = in[(threadIdx.x & 63) + 64];
    f64x4 acc0{0, 0, 0, 0}, acc1{1, 0, 0, 0}, acc2{2, 0, 0, 0}, acc3{3, 0, 0, 0};
    f64x4 acc4{0, 1, 0, 0}, acc5{0, 2, 0, 0}, acc6{0, 3, 0, 0}, acc7{0, 0, 1, 0};
    for (int i = 0; i < iters; ++i) {
        acc0 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc1, 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc2, 0, 0, 0);
        acc3 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc3, 0, 0, 0);
        acc4 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc4, 0, 0, 0);
        acc5 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc5, 0, 0, 0);
        acc6 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc6, 0, 0, 0);
        acc7 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc7, 0, 0, 0);
    }
    f64x4 s = acc0 + acc1 + acc2 + acc3 + acc4 + acc5 + acc6 + acc7;
    out[threadIdx.x + blockIdx.x * blockDim.x] = s[0] + s[1] + s[2] + s[3];
}

int main() {
    double *in, *out;
    (void)hipMalloc(&in, 128 * 8);
    (void)hipMalloc(&out, 256 * 2048 * 8);
    double h[128];
    for (int i = 0; i < 128; ++i) h[i] = 1.0 + i * 1e-3;  // random-ish, nonzero
    (void)hipMemcpy(in, h, sizeof h, hipMemcpyHostToDevice);
    const int iters = 20000;
    for (int rep = 0; rep < 3; ++rep) {
        for (int blocks : {256, 512, 1024, 2048}) {
            hipEvent_t a, b;
            (void)hipEventCreate(&a);
            (void)hipEventCreate(&b);
            (void)hipEventRecord(a, 0);
            hipLaunchKernelGGL(peak_kernel, dim3(blocks), dim3(256), 0, 0, in,
                               out, iters);
            (void)hipEventRecord(b, 0);
            (void)hipEventSynchronize(b);
            float ms = 0;
            (void)hipEventElapsedTime(&ms, a, b);
            const double flops = 8.0 * iters * 2048.0 * (blocks * 4.0);
            std::printf("rep %d blocks %4d: %7.2f TF/s\n", rep, blocks,
                        flops / (ms * 1e-3) / 1e12);
            (void)hipEventDestroy(a);
            (void)hipEventDestroy(b);
        }
    }
    return 0;
}
