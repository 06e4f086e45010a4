// Round-2 probe: grouped (MoE expert) Q40 GEMV lane utilization.
//
// Production k_q40_gemv_grouped (dllama_kernels.hip) assigns 2 rows/wave
// and strides 16B block-pairs by the FULL 64-lane wave. At Qwen3-30B-A3B
// shapes that leaves most lanes idle:
//   w13 per expert: n=2048 -> nbp=32 block-pairs -> lanes 32..63 idle
//   w2  per expert: n= 768 -> nbp=12             -> lanes 12..63 idle (5.3x)
// which matches the observed ~5x-off-stream grouped GEMV time.
//
// v2 here tiles a wave into 64/LPP lane-groups, each owning its own 2-row
// pair (LPP = smallest power of two >= nbp, capped at 64): every lane busy,
// more rows (and loads) in flight per wave.
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/moe_gemv_probe.hip -o /tmp/moe
// Run (GPU box): /tmp/moe   (self-checks vs CPU, times both shapes)
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>
#include <random>

#define WAVE 64
#define QB 32

#define HIP_CHECK(x) do { hipError_t e_ = (x); if (e_ != hipSuccess) { \
    fprintf(stderr, "HIP error %s at %d\n", hipGetErrorString(e_), __LINE__); \
    exit(1); } } while (0)

__device__ __forceinline__ int q40_block_dot(const uint4 &wq, const int4 &x0,
                                             const int4 &x1) {
    // bytewise lo/hi nibbles vs the packed int8 activation halves
    const uint32_t wv[4] = {wq.x, wq.y, wq.z, wq.w};
    const int xl[4] = {x0.x, x0.y, x0.z, x0.w};
    const int xh[4] = {x1.x, x1.y, x1.z, x1.w};
    int acc = 0;
#pragma unroll
    for (int t = 0; t < 4; t++) {
        acc = __builtin_amdgcn_sdot4((int)(wv[t] & 0x0F0F0F0Fu), xl[t], acc, false);
        acc = __builtin_amdgcn_sdot4((int)((wv[t] >> 4) & 0x0F0F0F0Fu), xh[t], acc,
                                     false);
    }
    return acc;
}

// NOTE: the probe packs activations so that block j's elems 0..15 pair with
// lo nibbles and 16..31 with hi nibbles, interleaved per uint32 (elems
// 4t..4t+3 lo / 4t+16..4t+19 hi per dword t) — same wire math as the
// production kernel's layout.

template <int LPP>  // lanes per 2-row pair (64 = production behavior)
__global__ void k_grouped_v2(const uint8_t *__restrict__ qs,
                             const __half *__restrict__ scales,
                             const int8_t *__restrict__ xq,
                             const float *__restrict__ xs,
                             const float *__restrict__ xbs,
                             const int *__restrict__ expert_idx,
                             float *__restrict__ y,
                             int d, int n, int k_slots) {
    constexpr int NGRP = WAVE / LPP;
    const int wpb = blockDim.x / WAVE;
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int grp = lane / LPP;
    const int lane_in = lane % LPP;
    const int row0 = (blockIdx.x * wpb + wave) * 2 * NGRP + grp * 2;
    const int slot = blockIdx.y;
    if (row0 >= d) return;
    const int nb = n / QB;
    const int nbp = nb >> 1;
    const int e = expert_idx[slot];
    const int b = slot / k_slots;
    const int row1 = min(row0 + 1, d - 1);
    const uint4 *wrow0 = reinterpret_cast<const uint4 *>(
        qs + ((int64_t)e * d + row0) * (n >> 1));
    const uint4 *wrow1 = reinterpret_cast<const uint4 *>(
        qs + ((int64_t)e * d + row1) * (n >> 1));
    const __half *srow0 = scales + ((int64_t)e * d + row0) * nb;
    const __half *srow1 = scales + ((int64_t)e * d + row1) * nb;
    float acc0 = 0.0f, acc1 = 0.0f;
    for (int jp = lane_in; jp < nbp; jp += LPP) {
        const int j = jp << 1;
        const uint4 a0 = wrow0[j], a1 = wrow0[j + 1];
        const uint4 b0 = wrow1[j], b1 = wrow1[j + 1];
        const float2 sw0 = __half22float2(*reinterpret_cast<const __half2 *>(srow0 + j));
        const float2 sw1 = __half22float2(*reinterpret_cast<const __half2 *>(srow1 + j));
        const int4 *xr = reinterpret_cast<const int4 *>(xq + (int64_t)b * n) + j * 2;
        const int4 x0 = xr[0], x1 = xr[1], x2 = xr[2], x3 = xr[3];
        const float2 sx = *reinterpret_cast<const float2 *>(xs + (int64_t)b * nb + j);
        const float2 bsum = *reinterpret_cast<const float2 *>(xbs + (int64_t)b * nb + j);
        acc0 = fmaf(sw0.x * sx.x, (float)q40_block_dot(a0, x0, x1) - 8.0f * bsum.x, acc0);
        acc0 = fmaf(sw0.y * sx.y, (float)q40_block_dot(a1, x2, x3) - 8.0f * bsum.y, acc0);
        acc1 = fmaf(sw1.x * sx.x, (float)q40_block_dot(b0, x0, x1) - 8.0f * bsum.x, acc1);
        acc1 = fmaf(sw1.y * sx.y, (float)q40_block_dot(b1, x2, x3) - 8.0f * bsum.y, acc1);
    }
    if ((nb & 1) && lane_in == 0) {
        const int j = nb - 1;
        const int4 *xb = reinterpret_cast<const int4 *>(xq + (int64_t)b * n) + j * 2;
        acc0 = fmaf(__half2float(srow0[j]) * xs[(int64_t)b * nb + j],
                    (float)q40_block_dot(wrow0[j], xb[0], xb[1])
                    - 8.0f * xbs[(int64_t)b * nb + j], acc0);
        acc1 = fmaf(__half2float(srow1[j]) * xs[(int64_t)b * nb + j],
                    (float)q40_block_dot(wrow1[j], xb[0], xb[1])
                    - 8.0f * xbs[(int64_t)b * nb + j], acc1);
    }
#pragma unroll
    for (int o = LPP / 2; o > 0; o >>= 1) {
        acc0 += __shfl_down(acc0, o, WAVE);
        acc1 += __shfl_down(acc1, o, WAVE);
    }
    if (lane_in == 0) {
        y[(int64_t)slot * d + row0] = acc0;
        if (row0 + 1 < d) y[(int64_t)slot * d + row0 + 1] = acc1;
    }
}

// ----------------------------------------------------------------- host
static void pack_block(const float *src, int8_t *q, float *s, float *bs) {
    float amax = 0;
    for (int i = 0; i < QB; i++) amax = fmaxf(amax, fabsf(src[i]));
    const float d = amax / 127.0f;
    const float inv = d > 0 ? 1.0f / d : 0.0f;
    int sum = 0;
    int8_t tmp[QB];
    for (int i = 0; i < QB; i++) {
        tmp[i] = (int8_t)rintf(src[i] * inv);
        sum += tmp[i];
    }
    // interleave: dword t holds elems 4t..4t+3 (lo half) / 16+4t.. (hi half)
    for (int t = 0; t < 4; t++)
        for (int bpos = 0; bpos < 4; bpos++) {
            q[t * 4 + bpos] = tmp[t * 4 + bpos];
            q[16 + t * 4 + bpos] = tmp[16 + t * 4 + bpos];
        }
    *s = d;
    *bs = (float)sum;
}

int main() {
    std::mt19937 rng(5);
    std::normal_distribution<float> nd(0.0f, 0.5f);
    std::uniform_int_distribution<int> nib(0, 15);

    struct Shape { const char *name; int d, n, slots, k_slots, E; };
    // Qwen3-30B-A3B TP=1 decode shapes: 8 active experts
    const Shape shapes[] = {{"w13 (1536x2048, 8 slots)", 1536, 2048, 8, 8, 16},
                            {"w2  (2048x768, 8 slots)", 2048, 768, 8, 1, 16}};

    for (const auto &sh : shapes) {
        const int d = sh.d, n = sh.n, nb = n / QB, E = sh.E, S = sh.slots;
        std::vector<uint8_t> qs((size_t)E * d * n / 2);
        std::vector<__half> sw((size_t)E * d * nb);
        for (auto &v : qs) v = (uint8_t)(nib(rng) | (nib(rng) << 4));
        for (auto &v : sw) v = __float2half(0.01f + 0.0001f * (rng() % 100));
        const int nx = sh.k_slots == 1 ? S : 1;  // per-slot vs shared input
        std::vector<float> xf((size_t)nx * n);
        for (auto &v : xf) v = nd(rng);
        std::vector<int8_t> xq((size_t)nx * n);
        std::vector<float> xs((size_t)nx * nb), xbs((size_t)nx * nb);
        for (int r = 0; r < nx; r++)
            for (int j = 0; j < nb; j++)
                pack_block(&xf[(size_t)r * n + j * QB],
                           &xq[(size_t)r * n + j * QB],
                           &xs[(size_t)r * nb + j], &xbs[(size_t)r * nb + j]);
        std::vector<int> idx(S);
        for (int s = 0; s < S; s++) idx[s] = (s * 3) % E;

        // CPU reference from the packed int8 + nibbles (exact int math)
        std::vector<float> ref((size_t)S * d);
        for (int s = 0; s < S; s++) {
            const int e = idx[s], b = s / sh.k_slots;
            for (int m = 0; m < d; m++) {
                double acc = 0;
                for (int j = 0; j < nb; j++) {
                    int dot = 0, bsum = 0;
                    for (int i = 0; i < QB; i++) {
                        // element i of block j: byte i%16, nibble i/16
                        const uint8_t byte =
                            qs[((size_t)e * d + m) * (n / 2) + j * 16 + (i % 16)];
                        const int w = ((i < 16) ? (byte & 15) : (byte >> 4));
                        const int xv = xq[(size_t)b * n + j * QB +
                                          (i % 16) / 4 * 4 + (i % 4) +
                                          (i / 16) * 16];
                        dot += w * xv;
                        bsum += xv;
                    }
                    acc += (double)__half2float(sw[((size_t)e * d + m) * nb + j]) *
                           xs[(size_t)b * nb + j] * (dot - 8.0 * bsum);
                }
                ref[(size_t)s * d + m] = (float)acc;
            }
        }

        uint8_t *dqs; __half *dsw; int8_t *dxq; float *dxs, *dbs, *dy; int *didx;
        HIP_CHECK(hipMalloc(&dqs, qs.size()));
        HIP_CHECK(hipMalloc(&dsw, sw.size() * 2));
        HIP_CHECK(hipMalloc(&dxq, xq.size()));
        HIP_CHECK(hipMalloc(&dxs, xs.size() * 4));
        HIP_CHECK(hipMalloc(&dbs, xbs.size() * 4));
        HIP_CHECK(hipMalloc(&didx, S * 4));
        HIP_CHECK(hipMalloc(&dy, (size_t)S * d * 4));
        HIP_CHECK(hipMemcpy(dqs, qs.data(), qs.size(), hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dsw, sw.data(), sw.size() * 2, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dxq, xq.data(), xq.size(), hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dxs, xs.data(), xs.size() * 4, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dbs, xbs.data(), xbs.size() * 4, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(didx, idx.data(), S * 4, hipMemcpyHostToDevice));

        printf("== %s (active-expert weights %.1f MB)\n", sh.name,
               (double)S * d * (n / 2 + 2 * nb) / 1e6);
        std::vector<float> out((size_t)S * d);
        hipEvent_t e0, e1;
        HIP_CHECK(hipEventCreate(&e0));
        HIP_CHECK(hipEventCreate(&e1));
        auto bench = [&](const char *name, int lpp, auto launch) {
            HIP_CHECK(hipMemset(dy, 0, (size_t)S * d * 4));
            launch();
            HIP_CHECK(hipDeviceSynchronize());
            HIP_CHECK(hipMemcpy(out.data(), dy, out.size() * 4,
                                hipMemcpyDeviceToHost));
            double maxrel = 0;
            for (size_t i = 0; i < out.size(); i++)
                maxrel = std::max(maxrel,
                                  (double)fabsf(out[i] - ref[i]) /
                                      (fabs((double)ref[i]) + 1e-2));
            const int iters = 500;
            HIP_CHECK(hipEventRecord(e0));
            for (int i = 0; i < iters; i++) launch();
            HIP_CHECK(hipEventRecord(e1));
            HIP_CHECK(hipEventSynchronize(e1));
            float ms;
            HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
            const double us = ms * 1000.0 / iters;
            const double gb = (double)S * d * (n / 2.0 + 2.0 * nb) / 1e9;
            printf("  LPP=%2d %s: %6.2f us  relerr %.2e %s  (%.2f TB/s)\n",
                   lpp, name, us, maxrel, maxrel < 2e-3 ? "OK" : "FAIL",
                   gb / (us * 1e-6) / 1000.0);
        };
        auto grid = [&](int lpp) {
            const int rows_per_wg = 4 * 2 * (WAVE / lpp);
            return dim3((d + rows_per_wg - 1) / rows_per_wg, S);
        };
        bench("production-equiv", 64, [&] {
            hipLaunchKernelGGL((k_grouped_v2<64>), grid(64), dim3(256), 0, 0,
                               dqs, dsw, dxq, dxs, dbs, didx, dy, d, n, sh.k_slots);
        });
        bench("lane-tiled      ", 32, [&] {
            hipLaunchKernelGGL((k_grouped_v2<32>), grid(32), dim3(256), 0, 0,
                               dqs, dsw, dxq, dxs, dbs, didx, dy, d, n, sh.k_slots);
        });
        bench("lane-tiled      ", 16, [&] {
            hipLaunchKernelGGL((k_grouped_v2<16>), grid(16), dim3(256), 0, 0,
                               dqs, dsw, dxq, dxs, dbs, didx, dy, d, n, sh.k_slots);
        });
        bench("lane-tiled      ", 8, [&] {
            hipLaunchKernelGGL((k_grouped_v2<8>), grid(8), dim3(256), 0, 0,
                               dqs, dsw, dxq, dxs, dbs, didx, dy, d, n, sh.k_slots);
        });
        HIP_CHECK(hipFree(dqs)); HIP_CHECK(hipFree(dsw)); HIP_CHECK(hipFree(dxq));
        HIP_CHECK(hipFree(dxs)); HIP_CHECK(hipFree(dbs)); HIP_CHECK(hipFree(didx));
        HIP_CHECK(hipFree(dy));
    }
    return 0;
}
