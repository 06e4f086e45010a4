// Round-2 probe: f16 KV cache for split-K decode attention (NOT runtime code).
//
// Production attention (k_attn_split/combine in dllama_kernels.hip) reads
// f32 KV — parity with the reference's f32 cache (nn-core.cpp:211-218).
// At long positions the decode attention is KV-bandwidth-bound (17 us at
// pos 4096, profiles/r01 notes); an f16 cache halves that traffic.
// This probe times a simplified split-K flash decode (B=1, online softmax,
// GQA) with f32 vs f16 KV and self-checks both against a CPU reference,
// quantifying the win and the numeric error before the production kernels
// grow a KV-dtype template.
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/attn_kv16_probe.hip -o /tmp/attn16
// Run (GPU box): /tmp/attn16 [n_heads kv_heads head_dim]
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>
#include <random>

#define WAVE 64

#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    exit(1); } } while (0)

__device__ __forceinline__ float wave_sum(float v) {
    for (int o = 32; o > 0; o >>= 1) v += __shfl_down(v, o, WAVE);
    return __shfl(v, 0, WAVE);
}
__device__ __forceinline__ float wave_max(float v) {
    for (int o = 32; o > 0; o >>= 1) v = fmaxf(v, __shfl_down(v, o, WAVE));
    return __shfl(v, 0, WAVE);
}

template <typename KVT>
__device__ __forceinline__ float kv_load(const KVT *p);
template <> __device__ __forceinline__ float kv_load<float>(const float *p) {
    return *p;
}
template <> __device__ __forceinline__ float kv_load<__half>(const __half *p) {
    return __half2float(*p);
}

// one workgroup (4 waves) per (head, split); each wave owns every 4th
// timestep of the split's range; online max/sum-rescaled accumulation,
// per-wave partials combined in LDS, per-split (m, l, o) written out.
template <typename KVT, int HD>
__global__ void k_attn_split_p(const float *__restrict__ q,   // [H, HD]
                               const KVT *__restrict__ kc,    // [T, KVH*HD]
                               const KVT *__restrict__ vc,
                               float *__restrict__ part,      // [H, S, 2+HD]
                               int n_heads, int kv_mul, int kv_dim,
                               int pos, int splits) {
    const int h = blockIdx.x;
    const int s = blockIdx.y;
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int steps = pos + 1;
    const int t0 = (int)((int64_t)steps * s / splits);
    const int t1 = (int)((int64_t)steps * (s + 1) / splits);
    const int kvh = h / kv_mul;
    const float scale = rsqrtf((float)HD);

    // q in registers: HD/WAVE elems per lane
    constexpr int E = HD / WAVE;
    float qr[E];
#pragma unroll
    for (int e = 0; e < E; e++) qr[e] = q[h * HD + lane + e * WAVE] * scale;

    float m = -1e30f, l = 0.0f, acc[E];
#pragma unroll
    for (int e = 0; e < E; e++) acc[e] = 0.0f;

    for (int t = t0 + wave; t < t1; t += 4) {
        const KVT *krow = kc + (int64_t)t * kv_dim + kvh * HD;
        float dot = 0.0f;
#pragma unroll
        for (int e = 0; e < E; e++)
            dot += qr[e] * kv_load<KVT>(krow + lane + e * WAVE);
        dot = wave_sum(dot);
        const float mn = fmaxf(m, dot);
        const float corr = __expf(m - mn);
        const float w = __expf(dot - mn);
        l = l * corr + w;
        const KVT *vrow = vc + (int64_t)t * kv_dim + kvh * HD;
#pragma unroll
        for (int e = 0; e < E; e++)
            acc[e] = acc[e] * corr + w * kv_load<KVT>(vrow + lane + e * WAVE);
        m = mn;
    }

    // combine the 4 waves' (m, l, acc) in LDS
    __shared__ float sm[4], sl[4], so[4][HD];
    if (lane < 1) { sm[wave] = m; sl[wave] = l; }
#pragma unroll
    for (int e = 0; e < E; e++) so[wave][lane + e * WAVE] = acc[e];
    __syncthreads();
    if (wave == 0) {
        float M = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
        float L = 0.0f, o[E];
#pragma unroll
        for (int e = 0; e < E; e++) o[e] = 0.0f;
        for (int wv = 0; wv < 4; wv++) {
            const float c = __expf(sm[wv] - M);
            L += sl[wv] * c;
#pragma unroll
            for (int e = 0; e < E; e++) o[e] += so[wv][lane + e * WAVE] * c;
        }
        float *dst = part + ((int64_t)h * gridDim.y + s) * (2 + HD);
        if (lane == 0) { dst[0] = M; dst[1] = L; }
#pragma unroll
        for (int e = 0; e < E; e++) dst[2 + lane + e * WAVE] = o[e];
    }
}

template <int HD>
__global__ void k_attn_combine_p(const float *__restrict__ part,
                                 float *__restrict__ out,
                                 int splits) {
    const int h = blockIdx.x;
    const int lane = threadIdx.x;
    float M = -1e30f;
    for (int s = 0; s < splits; s++)
        M = fmaxf(M, part[((int64_t)h * splits + s) * (2 + HD)]);
    float L = 0.0f;
    constexpr int E = HD / WAVE;
    float o[E];
#pragma unroll
    for (int e = 0; e < E; e++) o[e] = 0.0f;
    for (int s = 0; s < splits; s++) {
        const float *p = part + ((int64_t)h * splits + s) * (2 + HD);
        const float c = __expf(p[0] - M);
        L += p[1] * c;
#pragma unroll
        for (int e = 0; e < E; e++) o[e] += p[2 + lane + e * WAVE] * c;
    }
#pragma unroll
    for (int e = 0; e < E; e++)
        out[(int64_t)h * HD + lane + e * WAVE] = o[e] / L;
}

int main(int argc, char **argv) {
    const int H = argc > 1 ? atoi(argv[1]) : 32;
    const int KVH = argc > 2 ? atoi(argv[2]) : 8;
    const int HD = 128;
    const int kv_mul = H / KVH, kv_dim = KVH * HD;
    const int T = 4097;
    printf("attn probe H=%d KVH=%d HD=%d\n", H, KVH, HD);

    std::mt19937 rng(3);
    std::normal_distribution<float> nd(0.0f, 1.0f);
    std::vector<float> q(H * HD), kc((size_t)T * kv_dim), vc((size_t)T * kv_dim);
    for (auto &v : q) v = nd(rng);
    for (auto &v : kc) v = nd(rng) * 0.5f;
    for (auto &v : vc) v = nd(rng) * 0.5f;
    std::vector<__half> kh(kc.size()), vh(vc.size());
    for (size_t i = 0; i < kc.size(); i++) kh[i] = __float2half(kc[i]);
    for (size_t i = 0; i < vc.size(); i++) vh[i] = __float2half(vc[i]);

    float *dq, *dk32, *dv32, *dpart, *dout;
    __half *dk16, *dv16;
    HIP_CHECK(hipMalloc(&dq, q.size() * 4));
    HIP_CHECK(hipMalloc(&dk32, kc.size() * 4));
    HIP_CHECK(hipMalloc(&dv32, vc.size() * 4));
    HIP_CHECK(hipMalloc(&dk16, kh.size() * 2));
    HIP_CHECK(hipMalloc(&dv16, vh.size() * 2));
    HIP_CHECK(hipMalloc(&dpart, (size_t)H * 64 * (2 + HD) * 4));
    HIP_CHECK(hipMalloc(&dout, (size_t)H * HD * 4));
    HIP_CHECK(hipMemcpy(dq, q.data(), q.size() * 4, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dk32, kc.data(), kc.size() * 4, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dv32, vc.data(), vc.size() * 4, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dk16, kh.data(), kh.size() * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dv16, vh.data(), vh.size() * 2, hipMemcpyHostToDevice));

    std::vector<float> ref(H * HD), out(H * HD);
    auto cpu_ref = [&](int pos) {
        for (int h = 0; h < H; h++) {
            const int kvh = h / kv_mul;
            std::vector<double> sc(pos + 1);
            double mx = -1e30;
            for (int t = 0; t <= pos; t++) {
                double d = 0;
                for (int e = 0; e < HD; e++)
                    d += (double)q[h * HD + e] * kc[(size_t)t * kv_dim + kvh * HD + e];
                sc[t] = d / sqrt((double)HD);
                mx = std::max(mx, sc[t]);
            }
            double sum = 0;
            for (int t = 0; t <= pos; t++) { sc[t] = exp(sc[t] - mx); sum += sc[t]; }
            for (int e = 0; e < HD; e++) {
                double a = 0;
                for (int t = 0; t <= pos; t++)
                    a += sc[t] * vc[(size_t)t * kv_dim + kvh * HD + e];
                ref[h * HD + e] = (float)(a / sum);
            }
        }
    };

    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    for (int pos : {511, 1023, 2047, 4095}) {
        cpu_ref(pos);
        for (int S : {8, 16}) {
            auto run = [&](auto kvtag, const char *name, auto *kp, auto *vp) {
                using KVT = decltype(kvtag);
                hipLaunchKernelGGL((k_attn_split_p<KVT, 128>), dim3(H, S),
                                   dim3(256), 0, 0, dq, kp, vp, dpart,
                                   H, kv_mul, kv_dim, pos, S);
                hipLaunchKernelGGL((k_attn_combine_p<128>), dim3(H), dim3(WAVE),
                                   0, 0, dpart, dout, S);
                HIP_CHECK(hipDeviceSynchronize());
                HIP_CHECK(hipMemcpy(out.data(), dout, out.size() * 4,
                                    hipMemcpyDeviceToHost));
                double maxerr = 0;
                for (size_t i = 0; i < out.size(); i++)
                    maxerr = std::max(maxerr, (double)fabsf(out[i] - ref[i]));
                const int iters = 500;
                HIP_CHECK(hipEventRecord(e0));
                for (int i = 0; i < iters; i++) {
                    hipLaunchKernelGGL((k_attn_split_p<KVT, 128>), dim3(H, S),
                                       dim3(256), 0, 0, dq, kp, vp, dpart,
                                       H, kv_mul, kv_dim, pos, S);
                    hipLaunchKernelGGL((k_attn_combine_p<128>), dim3(H),
                                       dim3(WAVE), 0, 0, dpart, dout, S);
                }
                HIP_CHECK(hipEventRecord(e1));
                HIP_CHECK(hipEventSynchronize(e1));
                float ms;
                HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
                const double us = ms * 1000.0 / iters;
                // unique KV bytes (GQA re-reads come from L2)
                const double gb = 2.0 * (pos + 1) * kv_dim * sizeof(KVT) / 1e9;
                printf("pos=%4d S=%2d %s: %7.2f us  maxerr %.2e  (%.0f GB/s KV)\n",
                       pos, S, name, us, maxerr, gb / (us * 1e-6));
            };
            run(0.0f, "f32-kv", dk32, dv32);
            run(__half(), "f16-kv", dk16, dv16);
        }
    }
    return 0;
}
