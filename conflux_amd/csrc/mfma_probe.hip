// mfma_probe — empirically derive the v_mfma_f64_16x16x4_f64 operand and
// accumulator lane->element mappings on gfx950.  Debug tool, not product.
//   hipcc --offload-arch=gfx950 -O2 mfma_probe.hip -o mfma_probe && ./mfma_probe
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

typedef double f64x4 __attribute__((ext_vector_type(4)));

// Assumed mapping (guide's f32 16x16x4 analog):
//   a (1 double): A[lane&15][lane>>4]   (M=16 rows, K=4)
//   b (1 double): B[lane>>4][lane&15]   (K=4, N=16)
//   d[i]: D[(lane>>4)*4 + i][lane&15]
__global__ void probe(const double *A, const double *B, double *Draw) {
    const int l = threadIdx.x;
    const double a = A[(l & 15) * 4 + (l >> 4)];
    const double b = B[(l >> 4) * 16 + (l & 15)];
    f64x4 acc{0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
    for (int i = 0; i < 4; ++i) Draw[l * 4 + i] = acc[i];
}

int main() {
    double hA[16 * 4], hB[4 * 16], hD[16 * 16], hRaw[64 * 4];
    for (int i = 0; i < 64; ++i) hA[i] = i + 1;
    for (int i = 0; i < 64; ++i) hB[i] = (i + 1) * 0.001;
    // reference D = A(16x4) @ B(4x16)
    for (int r = 0; r < 16; ++r)
        for (int c = 0; c < 16; ++c) {
            double s = 0;
            for (int k = 0; k < 4; ++k) s += hA[r * 4 + k] * hB[k * 16 + c];
            hD[r * 16 + c] = s;
        }
    double *dA, *dB, *dR;
    (void)hipMalloc(&dA, sizeof hA);
    (void)hipMalloc(&dB, sizeof hB);
    (void)hipMalloc(&dR, sizeof hRaw);
    (void)hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
    (void)hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dR);
    (void)hipDeviceSynchronize();
    (void)hipMemcpy(hRaw, dR, sizeof hRaw, hipMemcpyDeviceToHost);

    // locate each (lane, i) value in D (all values distinct)
    int ok = 0, bad = 0;
    for (int l = 0; l < 64; ++l)
        for (int i = 0; i < 4; ++i) {
            const double x = hRaw[l * 4 + i];
            int fr = -1, fc = -1;
            for (int r = 0; r < 16 && fr < 0; ++r)
                for (int c = 0; c < 16; ++c)
                    if (fabs(hD[r * 16 + c] - x) < 1e-9) { fr = r; fc = c; break; }
            const int er = i * 4 + (l >> 4), ec = l & 15;  // verified D map
            if (fr == er && fc == ec) ++ok;
            else {
                ++bad;
                if (bad <= 8)
                    std::printf("lane %2d reg %d -> D[%d][%d], assumed D[%d][%d]\n",
                                l, i, fr, fc, er, ec);
            }
        }
    std::printf("assumed C/D mapping: %d ok, %d mismatched\n", ok, bad);
    return bad != 0;
}
