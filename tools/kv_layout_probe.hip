// KV layout probe: the production attention reads 256 B head-slices out of
// 2 KB [seq, kv_dim] rows (strided); a head-major [kvh, seq, hd] layout
// makes each wave's 16-timestep round a contiguous 4 KB stream and gives
// GQA re-reads L2 locality. This times the same split-attention math under
// both layouts at decode shapes (H=32, KVH=8, hd=128, f16 KV).
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/kv_layout_probe.hip -o /tmp/kvl
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdlib>

#define WAVE 64
#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP err %s @%d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

__device__ __forceinline__ float g16sum(float v) {
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 16);
    return v;
}

// TRANS=false: kc/vc are [seq, kv_dim] (production). TRANS=true: [kvh, seq, hd].
template <bool TRANS>
__global__ void k_attn_probe(const float *__restrict__ q, int q_ld,
                             const __half *__restrict__ kc,
                             const __half *__restrict__ vc,
                             int plen, int seq, int n_heads0, int kv_mul,
                             int kv_dim0, float scale,
                             float *__restrict__ ml, float *__restrict__ osc) {
    constexpr int VEC = 2, VEC16 = 8;
    const int h0 = blockIdx.x;
    const int b = blockIdx.y;
    const int sp = blockIdx.z;
    const int S = gridDim.z;
    const int hd = VEC * WAVE;
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int lane16 = lane & 15;
    const int group = lane >> 4;
    const int kvh = h0 / kv_mul;
    const int kv_off = TRANS ? 0 : kvh * hd;

    auto krow_at = [&](int t) {
        return TRANS ? kc + ((int64_t)kvh * seq + t) * hd + lane16 * VEC16
                     : kc + (int64_t)t * kv_dim0 + kv_off + lane16 * VEC16;
    };
    auto vrow_at = [&](int t) {
        return TRANS ? vc + ((int64_t)kvh * seq + t) * hd + lane * VEC
                     : vc + (int64_t)t * kv_dim0 + kv_off + lane * VEC;
    };

    float qreg[VEC16];
    #pragma unroll
    for (int v = 0; v < VEC16; v++)
        qreg[v] = q[(int64_t)b * q_ld + h0 * hd + lane16 * VEC16 + v] * scale;
    float m = -1e30f, l = 0.0f, o[VEC] = {0.0f, 0.0f};

    const int stride = 16 * S;
    for (int tb0 = (sp * 4 + wave) * 4; tb0 < plen; tb0 += 4 * stride) {
        float su[4];
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            const int tg = tb0 + u * stride + group;
            float partial = 0.0f;
            if (tg < plen) {
                const __half *kr = krow_at(tg);
                #pragma unroll
                for (int v = 0; v < VEC16; v++)
                    partial = fmaf(qreg[v], __half2float(kr[v]), partial);
            }
            su[u] = g16sum(partial);
        }
        float s16[16], w16[16];
        float mn = m;
        #pragma unroll
        for (int u = 0; u < 4; u++)
            #pragma unroll
            for (int gg = 0; gg < 4; gg++) {
                float v = __shfl(su[u], gg * 16, WAVE);
                if (tb0 + u * stride + gg >= plen) v = -1e30f;
                s16[4 * u + gg] = v;
                mn = fmaxf(mn, v);
            }
        const float f = __expf(m - mn);
        float ls = 0.0f;
        #pragma unroll
        for (int i = 0; i < 16; i++) { w16[i] = __expf(s16[i] - mn); ls += w16[i]; }
        l = l * f + ls;
        o[0] *= f; o[1] *= f;
        #pragma unroll
        for (int u = 0; u < 4; u++)
            #pragma unroll
            for (int gg = 0; gg < 4; gg++) {
                const int t = tb0 + u * stride + gg;
                if (t >= plen) continue;
                const __half *vr = vrow_at(t);
                o[0] = fmaf(w16[4 * u + gg], __half2float(vr[0]), o[0]);
                o[1] = fmaf(w16[4 * u + gg], __half2float(vr[1]), o[1]);
            }
        m = mn;
    }
    __shared__ float sm[4], sl[4];
    __shared__ float so[4][128];
    if (lane == 0) { sm[wave] = m; sl[wave] = l; }
    __syncthreads();
    const float M = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    const float fw = __expf(m - M);
    so[wave][lane * VEC] = o[0] * fw;
    so[wave][lane * VEC + 1] = o[1] * fw;
    __syncthreads();
    const int64_t slot = ((int64_t)b * n_heads0 + h0) * S + sp;
    if (wave == 0) {
        const float L = sl[0] * __expf(sm[0] - M) + sl[1] * __expf(sm[1] - M)
                      + sl[2] * __expf(sm[2] - M) + sl[3] * __expf(sm[3] - M);
        if (lane == 0) { ml[slot * 2] = M; ml[slot * 2 + 1] = L; }
        #pragma unroll
        for (int v = 0; v < VEC; v++) {
            const int i = lane * VEC + v;
            osc[slot * hd + i] = so[0][i] + so[1][i] + so[2][i] + so[3][i];
        }
    }
}

int main() {
    const int H = 32, KVH = 8, hd = 128, seq = 8192;
    const int kv_dim = KVH * hd;
    __half *kc, *vc;
    float *q, *ml, *osc;
    HIP_CHECK(hipMalloc(&kc, (size_t)seq * kv_dim * 2));
    HIP_CHECK(hipMalloc(&vc, (size_t)seq * kv_dim * 2));
    HIP_CHECK(hipMalloc(&q, H * hd * 4));
    HIP_CHECK(hipMalloc(&ml, H * 64 * 2 * 4));
    HIP_CHECK(hipMalloc(&osc, (size_t)H * 64 * hd * 4));
    HIP_CHECK(hipMemset(kc, 0x11, (size_t)seq * kv_dim * 2));
    HIP_CHECK(hipMemset(vc, 0x11, (size_t)seq * kv_dim * 2));
    const float scale = 1.0f / sqrtf((float)hd);
    printf("KV layout probe H=%d KVH=%d hd=%d (f16)\n", H, KVH, hd);
    for (int plen : {512, 1024, 2048, 4096}) {
        for (int S : {8, 16}) {
            for (int trans = 0; trans < 2; trans++) {
                hipEvent_t a, b2;
                HIP_CHECK(hipEventCreate(&a));
                HIP_CHECK(hipEventCreate(&b2));
                auto launch = [&]() {
                    if (trans)
                        hipLaunchKernelGGL(k_attn_probe<true>, dim3(H, 1, S),
                                           dim3(256), 0, 0, q, H * hd, kc, vc,
                                           plen, seq, H, H / KVH, kv_dim,
                                           scale, ml, osc);
                    else
                        hipLaunchKernelGGL(k_attn_probe<false>, dim3(H, 1, S),
                                           dim3(256), 0, 0, q, H * hd, kc, vc,
                                           plen, seq, H, H / KVH, kv_dim,
                                           scale, ml, osc);
                };
                launch();
                HIP_CHECK(hipDeviceSynchronize());
                HIP_CHECK(hipEventRecord(a));
                for (int i = 0; i < 200; i++) launch();
                HIP_CHECK(hipEventRecord(b2));
                HIP_CHECK(hipDeviceSynchronize());
                float ms;
                HIP_CHECK(hipEventElapsedTime(&ms, a, b2));
                const double us = ms * 1000.0 / 200;
                const double mb = (double)plen * hd * 2 * 2 * KVH / 1e6;
                printf("  plen=%4d S=%2d %s: %7.2f us (KV once: %5.1f MB, %4.0f GB/s)\n",
                       plen, S, trans ? "[kvh,seq,hd]" : "[seq,kvdim] ", us,
                       mb, mb * 1e3 / us);
            }
        }
    }
    return 0;
}
