// Workgroup-shape probe for round-2 fusion decisions (see profiles/r02):
// the fused W1|W3+SwiGLU kernel (16-wave wgs, one Q80 block per wg) measured
// 18.0 us vs 16.3 us for the unfused pair at Llama-8B shapes, and MoE
// consumers with per-wave gate recompute gained ~8 us each. This probe
// isolates WHY, by timing the same Q40 GEMV work under different wg shapes:
//
//   A. production shape: 4 waves/wg, RPW=2 (8 rows/wg)    [the r01 kernel]
//   B. 16 waves/wg, RPW=4 (64 rows/wg, one i-block/wg)    [fused-swiglu wg]
//   C. 8 waves/wg, RPW=8 (64 rows/wg)                     [half the waves]
//   D. 4 waves/wg, RPW=2, 512-thread... (8w RPW2, 16 rows)
//   E. moe_gate_wave VALU cost: gate vs no-gate prologue on shape A
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/fuse_shape_probe.hip -o /tmp/fsp
// Run:   /tmp/fsp [d n iters]
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define WAVE 64
#define QB 32
#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

__device__ __forceinline__ int q40_block_dot(const uint4 wq, const int4 x0,
                                             const int4 x1) {
    const uint32_t wv[4] = {wq.x, wq.y, wq.z, wq.w};
    const int32_t xv[8] = {x0.x, x0.y, x0.z, x0.w, x1.x, x1.y, x1.z, x1.w};
    int idot = 0;
    #pragma unroll
    for (int p = 0; p < 4; p++) {
        const uint32_t lo = wv[p] & 0x0F0F0F0Fu;
        const uint32_t hi = (wv[p] >> 4) & 0x0F0F0F0Fu;
        idot = __builtin_amdgcn_sdot4((int)lo, xv[2 * p], idot, false);
        idot = __builtin_amdgcn_sdot4((int)hi, xv[2 * p + 1], idot, false);
    }
    return idot;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
    return v;
}

// generic: WPB waves per wg, RPW rows per wave; wg covers WPB*RPW rows
template <int WPB, int RPW>
__global__ __launch_bounds__(WPB * WAVE) void k_gemv_shape(
        const uint8_t *__restrict__ qs, const __half *__restrict__ scales,
        const int8_t *__restrict__ xq, const float *__restrict__ xs,
        const float *__restrict__ xbs, float *__restrict__ y, int d, int n) {
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int row0 = (blockIdx.x * WPB + wid) * RPW;
    if (row0 >= d) return;
    const int nb = n / QB, nbp = nb >> 1;
    const uint4 *wrow[RPW];
    const __half *srow[RPW];
    #pragma unroll
    for (int r = 0; r < RPW; r++) {
        const int row = min(row0 + r, d - 1);
        wrow[r] = reinterpret_cast<const uint4 *>(qs + (int64_t)row * (n >> 1));
        srow[r] = scales + (int64_t)row * nb;
    }
    float acc[RPW];
    #pragma unroll
    for (int r = 0; r < RPW; r++) acc[r] = 0.0f;
    for (int jp = lane; jp < nbp; jp += WAVE) {
        const int j = jp << 1;
        const int4 *xr = reinterpret_cast<const int4 *>(xq) + j * 2;
        const int4 x0 = xr[0], x1 = xr[1], x2 = xr[2], x3 = xr[3];
        const float2 sx = *reinterpret_cast<const float2 *>(xs + j);
        const float2 bs = *reinterpret_cast<const float2 *>(xbs + j);
        #pragma unroll
        for (int r = 0; r < RPW; r++) {
            const uint4 w0 = wrow[r][j], w1 = wrow[r][j + 1];
            const float2 sw = __half22float2(*reinterpret_cast<const __half2 *>(srow[r] + j));
            acc[r] = fmaf(sw.x * sx.x, (float)q40_block_dot(w0, x0, x1) - 8.0f * bs.x, acc[r]);
            acc[r] = fmaf(sw.y * sx.y, (float)q40_block_dot(w1, x2, x3) - 8.0f * bs.y, acc[r]);
        }
    }
    #pragma unroll
    for (int r = 0; r < RPW; r++) {
        const float v = wave_reduce_sum(acc[r]);
        if (lane == 0 && row0 + r < d) y[row0 + r] = v;
    }
}

// gate-cost probe: same work as a 128-expert top-8 gate prologue, then the
// shape-A GEMV body; measures the marginal wall cost of per-wave gate VALU
__device__ __forceinline__ void gate_like(const float *__restrict__ logits,
                                          int lane, float *out) {
    float v[2];
    float m = -1e30f;
    #pragma unroll
    for (int i = 0; i < 2; i++) {
        v[i] = logits[lane * 2 + i];
        m = fmaxf(m, v[i]);
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, WAVE));
    float sum = 0.0f;
    #pragma unroll
    for (int i = 0; i < 2; i++) { v[i] = __expf(v[i] - m); sum += v[i]; }
    sum = wave_reduce_sum(sum);
    float acc = 0.0f;
    #pragma unroll
    for (int t = 0; t < 8; t++) {
        float best = -1.0f;
        int bi = -1;
        #pragma unroll
        for (int i = 0; i < 2; i++)
            if (v[i] > best) { best = v[i]; bi = lane * 2 + i; }
        #pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            const float ob = __shfl_xor(best, off, WAVE);
            const int oi = __shfl_xor(bi, off, WAVE);
            if (ob > best || (ob == best && oi >= 0 && (bi < 0 || oi < bi))) {
                best = ob; bi = oi;
            }
        }
        acc += best;
        #pragma unroll
        for (int i = 0; i < 2; i++)
            if (lane * 2 + i == bi) v[i] = -1.0f;
    }
    *out = acc / sum;
}

template <bool GATE>
__global__ void k_gemv_gate(const uint8_t *__restrict__ qs,
                            const __half *__restrict__ scales,
                            const int8_t *__restrict__ xq,
                            const float *__restrict__ xs,
                            const float *__restrict__ xbs,
                            const float *__restrict__ logits,
                            float *__restrict__ y, int d, int n) {
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    float g = 1.0f;
    if (GATE) gate_like(logits, lane, &g);
    const int row0 = (blockIdx.x * 4 + wid) * 2;
    if (row0 >= d) return;
    const int nb = n / QB, nbp = nb >> 1;
    const uint4 *w0p = reinterpret_cast<const uint4 *>(qs + (int64_t)row0 * (n >> 1));
    const uint4 *w1p = reinterpret_cast<const uint4 *>(qs + (int64_t)(row0 + 1) * (n >> 1));
    const __half *s0 = scales + (int64_t)row0 * nb;
    const __half *s1 = scales + (int64_t)(row0 + 1) * nb;
    float a0 = 0.0f, a1 = 0.0f;
    for (int jp = lane; jp < nbp; jp += WAVE) {
        const int j = jp << 1;
        const int4 *xr = reinterpret_cast<const int4 *>(xq) + j * 2;
        const int4 x0 = xr[0], x1 = xr[1], x2 = xr[2], x3 = xr[3];
        const float2 sx = *reinterpret_cast<const float2 *>(xs + j);
        const float2 bs = *reinterpret_cast<const float2 *>(xbs + j);
        const uint4 wa = w0p[j], wb = w0p[j + 1], wc = w1p[j], wd = w1p[j + 1];
        const float2 swa = __half22float2(*reinterpret_cast<const __half2 *>(s0 + j));
        const float2 swb = __half22float2(*reinterpret_cast<const __half2 *>(s1 + j));
        a0 = fmaf(swa.x * sx.x, (float)q40_block_dot(wa, x0, x1) - 8.0f * bs.x, a0);
        a0 = fmaf(swa.y * sx.y, (float)q40_block_dot(wb, x2, x3) - 8.0f * bs.y, a0);
        a1 = fmaf(swb.x * sx.x, (float)q40_block_dot(wc, x0, x1) - 8.0f * bs.x, a1);
        a1 = fmaf(swb.y * sx.y, (float)q40_block_dot(wd, x2, x3) - 8.0f * bs.y, a1);
    }
    a0 = wave_reduce_sum(a0) * g;
    a1 = wave_reduce_sum(a1) * g;
    if (lane == 0) { y[row0] = a0; y[row0 + 1] = a1; }
}

static float time_kernel(void (*launch)(void *), void *arg, int iters) {
    hipEvent_t a, b;
    HIP_CHECK(hipEventCreate(&a));
    HIP_CHECK(hipEventCreate(&b));
    launch(arg);  // warm
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipEventRecord(a));
    for (int i = 0; i < iters; i++) launch(arg);
    HIP_CHECK(hipEventRecord(b));
    HIP_CHECK(hipDeviceSynchronize());
    float ms;
    HIP_CHECK(hipEventElapsedTime(&ms, a, b));
    return ms * 1000.0f / iters;
}

struct Ctx {
    uint8_t *qs;
    __half *sc;
    int8_t *xq;
    float *xs, *xbs, *y, *logits;
    int d, n;
};

int main(int argc, char **argv) {
    const int d = argc > 1 ? atoi(argv[1]) : 28672;
    const int n = argc > 2 ? atoi(argv[2]) : 4096;
    const int iters = argc > 3 ? atoi(argv[3]) : 200;
    Ctx c;
    c.d = d; c.n = n;
    HIP_CHECK(hipMalloc(&c.qs, (size_t)d * n / 2));
    HIP_CHECK(hipMalloc(&c.sc, (size_t)d * (n / QB) * 2));
    HIP_CHECK(hipMalloc(&c.xq, n));
    HIP_CHECK(hipMalloc(&c.xs, n / QB * 4));
    HIP_CHECK(hipMalloc(&c.xbs, n / QB * 4));
    HIP_CHECK(hipMalloc(&c.y, (size_t)d * 4));
    HIP_CHECK(hipMalloc(&c.logits, 128 * 4));
    HIP_CHECK(hipMemset(c.qs, 0x57, (size_t)d * n / 2));
    HIP_CHECK(hipMemset(c.xq, 3, n));
    const double mb = (double)d * n / 2 / 1e6;
    printf("shape probe d=%d n=%d (%.1f MB weights)\n", d, n, mb);

    auto report = [&](const char *name, float us) {
        printf("  %-28s %7.2f us  (%5.2f TB/s)\n", name, us, mb / us / 1e3);
    };
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_shape<4, 2>), dim3((c->d + 7) / 8), dim3(256), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->y, c->d, c->n); };
        report("A: 4w RPW2 (8 rows/wg)", time_kernel(l, &c, iters));
    }
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_shape<16, 4>), dim3((c->d + 63) / 64), dim3(1024), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->y, c->d, c->n); };
        report("B: 16w RPW4 (64 rows/wg)", time_kernel(l, &c, iters));
    }
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_shape<8, 8>), dim3((c->d + 63) / 64), dim3(512), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->y, c->d, c->n); };
        report("C: 8w RPW8 (64 rows/wg)", time_kernel(l, &c, iters));
    }
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_shape<8, 2>), dim3((c->d + 15) / 16), dim3(512), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->y, c->d, c->n); };
        report("D: 8w RPW2 (16 rows/wg)", time_kernel(l, &c, iters));
    }
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_shape<16, 2>), dim3((c->d + 31) / 32), dim3(1024), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->y, c->d, c->n); };
        report("E: 16w RPW2 (32 rows/wg)", time_kernel(l, &c, iters));
    }
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_gate<false>), dim3((c->d + 7) / 8), dim3(256), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->logits, c->y, c->d, c->n); };
        report("F: shape A, no gate", time_kernel(l, &c, iters));
    }
    {
        auto l = [](void *p) { Ctx *c = (Ctx *)p;
            hipLaunchKernelGGL((k_gemv_gate<true>), dim3((c->d + 7) / 8), dim3(256), 0, 0,
                               c->qs, c->sc, c->xq, c->xs, c->xbs, c->logits, c->y, c->d, c->n); };
        report("G: shape A + gate VALU", time_kernel(l, &c, iters));
    }
    return 0;
}
