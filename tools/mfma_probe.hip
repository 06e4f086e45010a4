// Fragment-layout probe for v_mfma_i32_32x32x32_i8 on gfx950.
// Hypothesis (extension of the documented bf16 32x32x16 layout):
//   A[m][k]: lane = (m & 31) | ((k >> 4) << 5), byte = k & 15
//   B[k][n]: lane = (n & 31) | ((k >> 4) << 5), byte = k & 15
//   C[m][n]: lane = (n & 31) | ((m >> 2 & 1) << 5)?  -- use documented:
//            col = lane & 31, row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5)
// Verified by comparing against a CPU i32 matmul on random data.
// Build: hipcc --offload-arch=gfx950 -O2 tools/mfma_probe.hip -o /tmp/mfma_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

typedef int v4i __attribute__((ext_vector_type(4)));
typedef int v16i __attribute__((ext_vector_type(16)));

__global__ void k_probe(const signed char *A, const signed char *B, int *C) {
    const int lane = threadIdx.x;
    v4i a, b;
    signed char *ab = reinterpret_cast<signed char *>(&a);
    signed char *bb = reinterpret_cast<signed char *>(&b);
    const int mn = lane & 31;
    const int khi = lane >> 5;
    for (int t = 0; t < 16; t++) {
        const int k = khi * 16 + t;
        ab[t] = A[mn * 32 + k];   // A[m][k] row-major
        bb[t] = B[k * 32 + mn];   // B[k][n] row-major
    }
    v16i acc = {};
    acc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a, b, acc, 0, 0, 0);
    for (int r = 0; r < 16; r++) {
        const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        const int col = lane & 31;
        C[row * 32 + col] = acc[r];
    }
}

int main() {
    signed char hA[32 * 32], hB[32 * 32];
    int hC[32 * 32], ref[32 * 32];
    srand(42);
    for (int i = 0; i < 32 * 32; i++) {
        hA[i] = (signed char)(rand() % 255 - 127);
        hB[i] = (signed char)(rand() % 255 - 127);
    }
    for (int m = 0; m < 32; m++)
        for (int n = 0; n < 32; n++) {
            int s = 0;
            for (int k = 0; k < 32; k++) s += (int)hA[m * 32 + k] * (int)hB[k * 32 + n];
            ref[m * 32 + n] = s;
        }
    signed char *dA, *dB;
    int *dC;
    hipMalloc(&dA, 1024); hipMalloc(&dB, 1024); hipMalloc(&dC, 4096);
    hipMemcpy(dA, hA, 1024, hipMemcpyHostToDevice);
    hipMemcpy(dB, hB, 1024, hipMemcpyHostToDevice);
    hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, dA, dB, dC);
    hipMemcpy(hC, dC, 4096, hipMemcpyDeviceToHost);
    int bad = 0;
    for (int i = 0; i < 1024 && bad < 8; i++)
        if (hC[i] != ref[i]) {
            printf("MISMATCH at m=%d n=%d got %d want %d\n", i / 32, i % 32, hC[i], ref[i]);
            bad++;
        }
    printf(bad ? "FAIL\n" : "PASS: i8 32x32x32 layout confirmed\n");
    return bad ? 1 : 0;
}
