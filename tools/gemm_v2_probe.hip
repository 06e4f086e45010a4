// Round-2 prefill-GEMM prototype probe (NOT part of the runtime build).
//
// Standalone: self-checks numerics vs a CPU int reference and times the
// current production kernel (v1, copied from dllama_kernels.hip:909) against
// the v2 prototype. Build:
//   hipcc --offload-arch=gfx950 -O3 tools/gemm_v2_probe.hip -o /tmp/gemm_v2
// Run (GPU box):
//   /tmp/gemm_v2 [d n batch]
//
// v1 diagnosis (profiles/r01_decode_llama31_8b.md): 126 VGPR + 30 AGPR ->
// 3 waves/SIMD, ~64 VALU ops of descale per MFMA, activations re-read from
// L2 by every m-tile wave => MfmaUtil 1.5%, neither BW- nor compute-bound.
//
// v2 changes:
//  - activation fragments + x-scales staged in LDS per 8-block chunk,
//    cooperatively loaded ONCE per workgroup and shared by all 4 waves
//    (v1: every wave pulled the same 16B fragment from L2 every block)
//  - weight tiles pipelined through NAMED cur/next scalars (an indexed
//    ring array lands in scratch: the final code object shows 80 B/lane
//    private even when -Rpass remarks claim zero spill)
//  - descale arranged as float2 pairs (C rows r,r+1 are adjacent batch
//    rows) so the compiler can form v_pk_fma_f32
//  - target < 100 VGPRs -> 5-6 waves/SIMD
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>
#include <random>

#define WAVE 64
#define QB 32

typedef int v4i32_t __attribute__((ext_vector_type(4)));
typedef int v16i32_t __attribute__((ext_vector_type(16)));

static __device__ __forceinline__ void nib_extract(const uint4 &wq, int khi,
                                                   v4i32_t &b) {
    const uint32_t wv[4] = {wq.x, wq.y, wq.z, wq.w};
#pragma unroll
    for (int t = 0; t < 4; t++) {
        uint32_t s = khi ? ((wv[t] >> 4) & 0x0F0F0F0Fu) : (wv[t] & 0x0F0F0F0Fu);
        s ^= 0x08080808u;                       // bytewise (nibble - 8)
        b[t] = (int)(s | (((s >> 3) & 0x01010101u) * 0xF0u));
    }
}

// ---------------------------------------------------------------- v1 (prod)
__global__ void __launch_bounds__(256)
k_gemm_v1(const uint8_t *__restrict__ qs, const __half *__restrict__ scales,
          const int8_t *__restrict__ xq, const float *__restrict__ xs,
          float *__restrict__ y, int d, int n, int batch) {
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int mbase = (blockIdx.x * 4 + wave) * 32;
    if (mbase >= d) return;
    const int nb = n / QB;
    const int j0 = (int)((int64_t)nb * blockIdx.y / gridDim.y);
    const int j1 = (int)((int64_t)nb * (blockIdx.y + 1) / gridDim.y);
    const int khi = lane >> 5;
    const int bcol = lane & 31;
    const int mcol = lane & 31;
    const int mrow = min(mbase + mcol, d - 1);
    const uint4 *wrow = reinterpret_cast<const uint4 *>(qs + (int64_t)mrow * (n >> 1));
    const __half *srow = scales + (int64_t)mrow * nb;

    float facc[16];
#pragma unroll
    for (int r = 0; r < 16; r++) facc[r] = 0.0f;
    int j = j0;
    for (; j + 1 < j1; j += 2) {
        const float sxv0 = (lane < 32) ? xs[(int64_t)(lane & 31) * nb + j] : 0.0f;
        const float sxv1 = (lane < 32) ? xs[(int64_t)(lane & 31) * nb + j + 1] : 0.0f;
        const uint4 wq0 = wrow[j];
        const uint4 wq1 = wrow[j + 1];
        v4i32_t a0 = *reinterpret_cast<const v4i32_t *>(
            xq + (int64_t)bcol * n + j * QB + khi * 16);
        v4i32_t a1 = *reinterpret_cast<const v4i32_t *>(
            xq + (int64_t)bcol * n + (j + 1) * QB + khi * 16);
        v4i32_t b0, b1;
        nib_extract(wq0, khi, b0);
        nib_extract(wq1, khi, b1);
        v16i32_t i0 = {}, i1 = {};
        i0 = __builtin_amdgcn_mfma_i32_32x32x32_i8(a0, b0, i0, 0, 0, 0);
        i1 = __builtin_amdgcn_mfma_i32_32x32x32_i8(a1, b1, i1, 0, 0, 0);
        const float sw0 = __half2float(srow[j]);
        const float sw1 = __half2float(srow[j + 1]);
#pragma unroll
        for (int r = 0; r < 16; r++) {
            const int brow = (r & 3) + 8 * (r >> 2) + 4 * khi;
            facc[r] = fmaf((float)i0[r], sw0 * __shfl(sxv0, brow, WAVE), facc[r]);
            facc[r] = fmaf((float)i1[r], sw1 * __shfl(sxv1, brow, WAVE), facc[r]);
        }
    }
    for (; j < j1; j++) {
        const float sxv = (lane < 32) ? xs[(int64_t)(lane & 31) * nb + j] : 0.0f;
        v4i32_t a = *reinterpret_cast<const v4i32_t *>(
            xq + (int64_t)bcol * n + j * QB + khi * 16);
        v4i32_t b;
        nib_extract(wrow[j], khi, b);
        v16i32_t iacc = {};
        iacc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a, b, iacc, 0, 0, 0);
        const float sw = __half2float(srow[j]);
#pragma unroll
        for (int r = 0; r < 16; r++) {
            const int brow = (r & 3) + 8 * (r >> 2) + 4 * khi;
            facc[r] = fmaf((float)iacc[r], sw * __shfl(sxv, brow, WAVE), facc[r]);
        }
    }
    if (mbase + mcol < d) {
#pragma unroll
        for (int r = 0; r < 16; r++) {
            const int brow = (r & 3) + 8 * (r >> 2) + 4 * khi;
            if (brow < batch)
                atomicAdd(&y[(int64_t)brow * d + mbase + mcol], facc[r]);
        }
    }
}

// ---------------------------------------------------------------- v2 proto
// 4 waves/wg, one 32-row m-tile per wave; activations + x-scales staged in
// LDS per CHUNK blocks; weights pipelined via named cur/next scalars.
#define CHUNK 8
typedef __attribute__((address_space(3))) void *lds_ptr_t;

// DMA=true stages fragments with global_load_lds_dwordx4 (direct
// global->LDS, no VGPR round-trip): one instruction moves 1 KB per wave
// (lane i's 16 B land at lds_base + i*16, so the source address is arranged
// so lane i reads batch row i/2, half i%2).
template <bool DMA>
__global__ void __launch_bounds__(256)
k_gemm_v2(const uint8_t *__restrict__ qs, const __half *__restrict__ scales,
          const int8_t *__restrict__ xq, const float *__restrict__ xs,
          float *__restrict__ y, int d, int n, int batch) {
    __shared__ int8_t lds_a[2][CHUNK][32][QB];  // [buf][block][batch][elem]
    __shared__ float lds_s[2][CHUNK][32];       // [buf][block][batch]
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int tid = threadIdx.x;
    const int mbase = (blockIdx.x * 4 + wave) * 32;
    const int nb = n / QB;
    const int j0 = (int)((int64_t)nb * blockIdx.y / gridDim.y);
    const int j1 = (int)((int64_t)nb * (blockIdx.y + 1) / gridDim.y);
    const int khi = lane >> 5;
    const int mcol = lane & 31;
    const int mrow = min(mbase + mcol, d - 1);
    const uint4 *wrow = reinterpret_cast<const uint4 *>(qs + (int64_t)mrow * (n >> 1));
    const __half *srow = scales + (int64_t)mrow * nb;
    const bool live = mbase < d;

    // cooperative stage of blocks [jc, jc+CHUNK) into buffer `buf`:
    // 8 KB of fragments = 32 B/thread (2x uint4), 256 scale floats
    auto stage = [&](int buf, int jc) {
        const int nblk = min(CHUNK, j1 - jc);
        if constexpr (DMA) {
            // wave w stages blocks w, w+4, ...: one dwordx4 LDS-DMA moves
            // the whole 1 KB fragment (lane i -> batch row i/2, half i%2)
            for (int blk = wave; blk < nblk; blk += 4) {
                const int8_t *src = xq + (int64_t)(lane >> 1) * n +
                                    (jc + blk) * QB + (lane & 1) * 16;
                __builtin_amdgcn_load_to_lds((void *)src,
                                             (lds_ptr_t)&lds_a[buf][blk][0][0],
                                             16, 0, 0);
            }
        } else {
            // 256 threads copy 32 batch rows x nblk blocks, 32 B each
            for (int u = tid; u < nblk * 32; u += 256) {
                const int b = u & 31;
                const int blk = u >> 5;
                *reinterpret_cast<uint4 *>(&lds_a[buf][blk][b][0]) =
                    *reinterpret_cast<const uint4 *>(xq + (int64_t)b * n + (jc + blk) * QB);
                *reinterpret_cast<uint4 *>(&lds_a[buf][blk][b][16]) =
                    *reinterpret_cast<const uint4 *>(xq + (int64_t)b * n + (jc + blk) * QB + 16);
            }
        }
        for (int u = tid; u < nblk * 32; u += 256) {
            const int b = u & 31;
            const int blk = u >> 5;
            lds_s[buf][blk][b] = xs[(int64_t)b * nb + (jc + blk)];
        }
    };

    float2 facc[8];
#pragma unroll
    for (int r = 0; r < 8; r++) facc[r] = make_float2(0.0f, 0.0f);

    stage(0, j0);
    // weight pipeline as NAMED scalars: an indexed ring array is allocated
    // to scratch (80 B/lane; the final code object shows it even when
    // -Rpass remarks claim 0 spill) — named uint4s cannot spill
    uint4 wq_cur = {};
    float sw_cur = 0.0f;
    if (live && j0 < j1) { wq_cur = wrow[j0]; sw_cur = __half2float(srow[j0]); }
    if constexpr (DMA) __builtin_amdgcn_s_waitcnt(0);  // LDS-DMA uses vmcnt
    __syncthreads();

    int buf = 0;
    for (int jc = j0; jc < j1; jc += CHUNK, buf ^= 1) {
        const int nblk = min(CHUNK, j1 - jc);
        // kick off the next chunk's staging; its ds_writes complete before
        // the barrier below
        if (jc + CHUNK < j1) stage(buf ^ 1, jc + CHUNK);
        if (live) {
#pragma unroll
            for (int jj = 0; jj < CHUNK; jj++) {
                if (jj >= nblk) break;
                const int j = jc + jj;
                const uint4 wq = wq_cur;
                const float sw = sw_cur;
                if (j + 1 < j1) {
                    wq_cur = wrow[j + 1];
                    sw_cur = __half2float(srow[j + 1]);
                }
                v4i32_t a = *reinterpret_cast<const v4i32_t *>(
                    &lds_a[buf][jj][lane & 31][khi * 16]);
                v4i32_t b;
                nib_extract(wq, khi, b);
                v16i32_t iacc = {};
                iacc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a, b, iacc, 0, 0, 0);
#pragma unroll
                for (int r2 = 0; r2 < 8; r2++) {
                    // C rows 2*r2, 2*r2+1 are adjacent batch rows; their two
                    // x-scales sit side by side in LDS -> one broadcast
                    // ds_read_b64 instead of two ds_bpermutes
                    const int brow = ((2 * r2) & 3) + 8 * (r2 >> 1) + 4 * khi;
                    const float2 sx2 = *reinterpret_cast<const float2 *>(
                        &lds_s[buf][jj][brow]);
                    const float2 m2 = make_float2(sw * sx2.x, sw * sx2.y);
                    facc[r2].x = fmaf((float)iacc[2 * r2], m2.x, facc[r2].x);
                    facc[r2].y = fmaf((float)iacc[2 * r2 + 1], m2.y, facc[r2].y);
                }
            }
        }
        if constexpr (DMA) __builtin_amdgcn_s_waitcnt(0);  // LDS-DMA uses vmcnt
        __syncthreads();
    }

    if (live && mbase + mcol < d) {
#pragma unroll
        for (int r2 = 0; r2 < 8; r2++) {
            const int brow = ((2 * r2) & 3) + 8 * (r2 >> 1) + 4 * khi;
            if (brow < batch)
                atomicAdd(&y[(int64_t)brow * d + mbase + mcol], facc[r2].x);
            if (brow + 1 < batch)
                atomicAdd(&y[(int64_t)(brow + 1) * d + mbase + mcol], facc[r2].y);
        }
    }
}

// ---------------------------------------------------------------- host
#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__, __LINE__); \
    exit(1); } } while (0)

int main(int argc, char **argv) {
    int d = argc > 1 ? atoi(argv[1]) : 14336;
    int n = argc > 2 ? atoi(argv[2]) : 4096;
    int batch = argc > 3 ? atoi(argv[3]) : 32;
    const int nb = n / QB;
    printf("GEMM probe d=%d n=%d batch=%d\n", d, n, batch);

    std::mt19937 rng(7);
    std::uniform_int_distribution<int> nibd(0, 15), i8d(-127, 127);
    std::uniform_real_distribution<float> fd(0.001f, 0.02f);

    std::vector<uint8_t> qs((size_t)d * n / 2);
    std::vector<__half> sw((size_t)d * nb);
    std::vector<int8_t> xq((size_t)32 * n, 0);
    std::vector<float> xsv((size_t)32 * nb, 0.0f);
    for (auto &v : qs) v = (uint8_t)(nibd(rng) | (nibd(rng) << 4));
    for (auto &v : sw) v = __float2half(fd(rng));
    for (int b = 0; b < batch; b++) {
        for (int i = 0; i < n; i++) xq[(size_t)b * n + i] = (int8_t)i8d(rng);
        for (int j = 0; j < nb; j++) xsv[(size_t)b * nb + j] = fd(rng);
    }

    // CPU reference (f64 accumulate of exact int dots)
    std::vector<float> ref((size_t)batch * d);
    for (int b = 0; b < batch; b++)
        for (int m = 0; m < d; m++) {
            double acc = 0.0;
            for (int j = 0; j < nb; j++) {
                int dot = 0;
                for (int k = 0; k < QB; k++) {
                    const uint8_t byte = qs[(size_t)m * (n / 2) + j * 16 + (k & 15)];
                    const int w = ((k < 16) ? (byte & 15) : (byte >> 4)) - 8;
                    dot += w * (int)xq[(size_t)b * n + j * QB + k];
                }
                acc += (double)__half2float(sw[(size_t)m * nb + j]) *
                       (double)xsv[(size_t)b * nb + j] * (double)dot;
            }
            ref[(size_t)b * d + m] = (float)acc;
        }

    uint8_t *dqs; __half *dsw; int8_t *dxq; float *dxs, *dy;
    HIP_CHECK(hipMalloc(&dqs, qs.size()));
    HIP_CHECK(hipMalloc(&dsw, sw.size() * 2));
    HIP_CHECK(hipMalloc(&dxq, xq.size()));
    HIP_CHECK(hipMalloc(&dxs, xsv.size() * 4));
    HIP_CHECK(hipMalloc(&dy, (size_t)batch * d * 4));
    HIP_CHECK(hipMemcpy(dqs, qs.data(), qs.size(), hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dsw, sw.data(), sw.size() * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dxq, xq.data(), xq.size(), hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dxs, xsv.data(), xsv.size() * 4, hipMemcpyHostToDevice));

    const int mtiles = (d + 127) / 128;
    int ksplit = 1;
    while (mtiles * ksplit < 1024 && ksplit < 16 && (nb / (ksplit * 2)) >= CHUNK)
        ksplit *= 2;
    printf("grid: %d x %d (ksplit)\n", mtiles, ksplit);

    std::vector<float> out((size_t)batch * d);
    auto check = [&](const char *name) {
        HIP_CHECK(hipMemcpy(out.data(), dy, out.size() * 4, hipMemcpyDeviceToHost));
        double maxrel = 0;
        for (size_t i = 0; i < out.size(); i++) {
            const double rel = fabs(out[i] - ref[i]) /
                               (fabs((double)ref[i]) + 1e-3);
            if (rel > maxrel) maxrel = rel;
        }
        printf("%s max rel err: %.3e %s\n", name, maxrel,
               maxrel < 2e-3 ? "OK" : "FAIL");
        return maxrel < 2e-3;
    };
    auto bench = [&](const char *name, auto launch) {
        HIP_CHECK(hipMemset(dy, 0, (size_t)batch * d * 4));
        launch();
        HIP_CHECK(hipDeviceSynchronize());
        if (!check(name)) return;
        const int iters = 200;
        hipEvent_t e0, e1;
        HIP_CHECK(hipEventCreate(&e0));
        HIP_CHECK(hipEventCreate(&e1));
        HIP_CHECK(hipEventRecord(e0));
        for (int i = 0; i < iters; i++) launch();
        HIP_CHECK(hipEventRecord(e1));
        HIP_CHECK(hipEventSynchronize(e1));
        float ms;
        HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
        const double us = ms * 1000.0 / iters;
        const double gb = ((double)d * n / 2 + (double)d * nb * 2) / 1e9;
        printf("%s: %.1f us  (%.2f TB/s weight stream, %.1f TFLOP int8)\n",
               name, us, gb / (us * 1e-6) / 1000.0,
               2.0 * d * n * batch / (us * 1e-6) / 1e12);
    };

    bench("v1", [&] {
        hipLaunchKernelGGL(k_gemm_v1, dim3(mtiles, ksplit), dim3(256), 0, 0,
                           dqs, dsw, dxq, dxs, dy, d, n, batch);
    });
    bench("v2", [&] {
        hipLaunchKernelGGL((k_gemm_v2<false>), dim3(mtiles, ksplit), dim3(256),
                           0, 0, dqs, dsw, dxq, dxs, dy, d, n, batch);
    });
    bench("v2+ldsdma", [&] {
        hipLaunchKernelGGL((k_gemm_v2<true>), dim3(mtiles, ksplit), dim3(256),
                           0, 0, dqs, dsw, dxq, dxs, dy, d, n, batch);
    });
    return 0;
}
