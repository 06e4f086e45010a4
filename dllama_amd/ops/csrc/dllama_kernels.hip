// dllama_amd — hand-written CDNA4 (gfx950) kernels for Q40xQ80 LLM inference.
//
// Design notes (MI355X-first, not a port):
//  - decode is HBM-bandwidth bound on the Q40 weight stream; the GEMV keeps
//    one wave per output row streaming 16B nibble payloads per lane and does
//    the int8 math with v_dot4 (sdot4), correcting the Q40 "-8" offset with
//    precomputed per-block activation sums (so nibbles never get unpacked
//    to signed values).
//  - every runtime scalar the decode loop needs (position) is read from
//    device memory so whole-token hipGraph capture works.
//  - role parity with the reference op set: src/nn/nn-cpu-ops.cpp and the
//    Vulkan shaders in src/nn/vulkan/ (see SURVEY.md §2.3/§2.5).
//
// All f32 accumulation (parity with the reference numerics).

#include <hip/hip_runtime.h>
#if defined(__AVX2__)
#include <immintrin.h>
#endif
#include <vector>
#include <thread>
#if defined(_OPENMP)
#include <omp.h>
#endif
#include <hip/hip_fp16.h>
#include <torch/extension.h>
#include <cstdlib>
#include <ATen/hip/HIPContext.h>

#define WAVE 64
#define QB 32  // quant block size
#define SSQ_SPREAD 16
#define SSQ_PAD 32  // floats between spread slots (own 128B cacheline)

static inline int ceil_div(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

#if defined(__HIPCC__)
#if __has_builtin(__builtin_amdgcn_sdot4)
__device__ __forceinline__ int dot4(int a, int b, int c) {
    return __builtin_amdgcn_sdot4(a, b, c, false);
}
#else
__device__ __forceinline__ int dot4(int a, int b, int c) {
    const char4 va = *reinterpret_cast<const char4 *>(&a);
    const char4 vb = *reinterpret_cast<const char4 *>(&b);
    return c + va.x * vb.x + va.y * vb.y + va.z * vb.z + va.w * vb.w;
}
#endif

__device__ __forceinline__ float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_xor(v, off, WAVE);
    return v;
}

__device__ __forceinline__ float group32_reduce_max(float v) {
    #pragma unroll
    for (int off = 16; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, 32));
    return v;
}

__device__ __forceinline__ float group32_reduce_sum(float v) {
    #pragma unroll
    for (int off = 16; off > 0; off >>= 1)
        v += __shfl_xor(v, off, 32);
    return v;
}

__device__ __forceinline__ float group16_reduce_sum(float v) {
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1)
        v += __shfl_xor(v, off, 16);
    return v;
}

__device__ __forceinline__ float group16_reduce_max(float v) {
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, 16));
    return v;
}

// whole-wave MoE gate (reference OP_SOFTMAX + OP_MOE_GATE,
// nn-cpu-ops.cpp:595-720,1462-1492): softmax over the router logits, then
// iterative top-k with smallest-index tie-break, weights normalized by the
// top-k sum. Every lane of the executing wave returns the full idx/weight
// arrays (static-indexed — runtime-indexed locals land in scratch).
// Callable from consumer kernels so the gate costs ~0.3 us of redundant
// VALU per wave instead of its own ~10 us single-workgroup launch.
__device__ __forceinline__ void moe_gate_wave(
        const float *__restrict__ logits, int n_experts, int topk, int lane,
        int *out_idx, float *out_w) {
    const int per = (n_experts + WAVE - 1) / WAVE;
    // fixed-trip unrolled loops: runtime-indexed arrays go to scratch
    float v[16];  // per-lane expert logits (supports n_experts <= 1024)
    float m = -1e30f;
    #pragma unroll
    for (int i = 0; i < 16; i++) {
        const int eidx = lane * per + i;
        v[i] = (i < per && eidx < n_experts) ? logits[eidx] : -1e30f;
        m = fmaxf(m, v[i]);
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off, WAVE));
    float sum = 0.0f;
    #pragma unroll
    for (int i = 0; i < 16; i++) {
        v[i] = (i < per && (lane * per + i) < n_experts) ? __expf(v[i] - m) : 0.0f;
        sum += v[i];
    }
    sum = wave_reduce_sum(sum);
    const float inv = 1.0f / sum;
    // iterative top-k: packed (prob, smallest-index-wins) max per round
    float wsum = 0.0f;
    float chosen[16];  // topk <= 16
    #pragma unroll
    for (int t = 0; t < 16; t++) {
        if (t >= topk) continue;  // guarded full unroll (break blocks unrolling)
        float best = -1.0f;
        int bi = -1;
        #pragma unroll
        for (int i = 0; i < 16; i++) {
            if (i < per && v[i] > best) { best = v[i]; bi = lane * per + i; }
        }
        // wave argmax: (prob, -index) lexicographic via packed compare
        #pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            const float ob = __shfl_xor(best, off, WAVE);
            const int oi = __shfl_xor(bi, off, WAVE);
            if (ob > best || (ob == best && oi >= 0 && (bi < 0 || oi < bi))) {
                best = ob; bi = oi;
            }
        }
        out_idx[t] = bi;
        chosen[t] = best * inv;
        wsum += best * inv;
        // clear the winner (static-index scan, see scratch note above)
        #pragma unroll
        for (int i = 0; i < 16; i++)
            if (lane * per + i == bi) v[i] = -1.0f;
    }
    // normalize by the top-k sum (reference normTopk)
    const float winv = 1.0f / wsum;
    #pragma unroll
    for (int t = 0; t < 16; t++)
        out_w[t] = t < topk ? chosen[t] * winv : 0.0f;
}


__device__ __forceinline__ float ssq_total(const float *ssq, int b) {
    float t = 0.0f;
    #pragma unroll
    for (int k = 0; k < SSQ_SPREAD; k++)
        t += ssq[(b * SSQ_SPREAD + k) * SSQ_PAD];
    return t;
}

// wave-parallel variant: the 16 spread slots live on 16 different
// cachelines (by design — atomics), so the serial loop above costs ~16
// dependent-ish loads at every wave's kernel start; here lanes 0..15 each
// load one slot and a 16-group reduce + broadcast distributes the sum
// (used by the PRO consumers, which run on every wave of 3.5k workgroups)
__device__ __forceinline__ float ssq_total_wave(const float *ssq, int b,
                                                int lane) {
    float t = (lane & 63) < SSQ_SPREAD
                  ? ssq[(b * SSQ_SPREAD + (lane & 15)) * SSQ_PAD] : 0.0f;
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1)
        t += __shfl_xor(t, off, 16);
    return __shfl(t, 0, WAVE);
}

// ------------------------------------------------------------------ q80 quantize
// f32 [rows, n] -> int8 q [rows, n], f32 scale [rows, n/32], f32 bsum [rows, n/32]
// (role of reference quantizeF32toQ80, nn-quants.cpp:67 and the
//  cast-forward-f32-q80 Vulkan shader). bsum = sum of the int8 values,
// used by the GEMV to fold the Q40 nibble offset.
__global__ void k_q80_quantize(const float *__restrict__ x,
                               int8_t *__restrict__ q,
                               float *__restrict__ s,
                               float *__restrict__ bs,
                               int n_blocks_total) {
    // one 32-lane group per block
    int gid = (blockIdx.x * blockDim.x + threadIdx.x) / 32;
    int lane = threadIdx.x & 31;
    if (gid >= n_blocks_total) return;
    float v = x[gid * QB + lane];
    float amax = group32_reduce_max(fabsf(v));
    float d = amax / 127.0f;
    float inv = d > 0.0f ? 1.0f / d : 0.0f;
    float qf = rintf(v * inv);
    int8_t qi = (int8_t)qf;
    q[gid * QB + lane] = qi;
    float bsum = group32_reduce_sum(qf);
    if (lane == 0) {
        s[gid] = d;
        bs[gid] = bsum;
    }
}

// ------------------------------------------------------------------ rmsnorm
// y = x * w / sqrt(mean(x^2)+eps) (reference invRms_F32 + rmsNorm_F32,
// nn-cpu-ops.cpp:114-175, fused). One workgroup per row.
__global__ void k_rmsnorm(const float *__restrict__ x,
                          const float *__restrict__ w,
                          float *__restrict__ y,
                          int n, float eps) {
    const float *row = x + (int64_t)blockIdx.x * n;
    float *out = y + (int64_t)blockIdx.x * n;
    float acc = 0.0f;
    for (int i = threadIdx.x; i < n; i += blockDim.x) {
        float v = row[i];
        acc += v * v;
    }
    __shared__ float red[16];
    acc = wave_reduce_sum(acc);
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = acc;
    __syncthreads();
    int nw = blockDim.x / WAVE;
    float total = 0.0f;
    for (int i = 0; i < nw; i++) total += red[i];
    float inv = rsqrtf(total / n + eps);
    for (int i = threadIdx.x; i < n; i += blockDim.x)
        out[i] = row[i] * inv * w[i];
}

// fused rmsnorm + q80 quantize: avoids a full extra activation pass
// (reference runs inv_rms -> rms_norm -> cast as three ops).
__global__ void k_rmsnorm_q80(const float *__restrict__ x,
                              const float *__restrict__ w,
                              int8_t *__restrict__ q,
                              float *__restrict__ s,
                              float *__restrict__ bs,
                              int n, float eps) {
    const float *row = x + (int64_t)blockIdx.x * n;
    float acc = 0.0f;
    for (int i = threadIdx.x; i < n; i += blockDim.x) {
        float v = row[i];
        acc += v * v;
    }
    __shared__ float red[16];
    acc = wave_reduce_sum(acc);
    int wid = threadIdx.x / WAVE, lane64 = threadIdx.x % WAVE;
    if (lane64 == 0) red[wid] = acc;
    __syncthreads();
    int nw = blockDim.x / WAVE;
    float total = 0.0f;
    for (int i = 0; i < nw; i++) total += red[i];
    float inv = rsqrtf(total / n + eps);

    int nb = n / QB;
    int8_t *qrow = q + (int64_t)blockIdx.x * n;
    float *srow = s + (int64_t)blockIdx.x * nb;
    float *bsrow = bs + (int64_t)blockIdx.x * nb;
    int lane32 = threadIdx.x & 31;
    for (int blk = threadIdx.x / 32; blk < nb; blk += blockDim.x / 32) {
        float v = row[blk * QB + lane32] * inv * w[blk * QB + lane32];
        float amax = group32_reduce_max(fabsf(v));
        float d = amax / 127.0f;
        float qinv = d > 0.0f ? 1.0f / d : 0.0f;
        float qf = rintf(v * qinv);
        qrow[blk * QB + lane32] = (int8_t)qf;
        float bsum = group32_reduce_sum(qf);
        if (lane32 == 0) { srow[blk] = d; bsrow[blk] = bsum; }
    }
}

// per-head rmsnorm (Qwen3 q/k-norm; reference multi-column OP_INV_RMS +
// OP_RMS_NORM, llm.cpp:178-187). x [rows, hd], w [hd]; hd <= 512.
__global__ void k_rmsnorm_rows(const float *__restrict__ x,
                               const float *__restrict__ w,
                               float *__restrict__ y,
                               int hd, float eps) {
    const float *row = x + (int64_t)blockIdx.x * hd;
    float *out = y + (int64_t)blockIdx.x * hd;
    float acc = 0.0f;
    for (int i = threadIdx.x; i < hd; i += blockDim.x) {
        float v = row[i];
        acc += v * v;
    }
    acc = wave_reduce_sum(acc);  // blockDim == 64 (one wave)
    float inv = rsqrtf(acc / hd + eps);
    for (int i = threadIdx.x; i < hd; i += blockDim.x)
        out[i] = row[i] * inv * w[i];
}

// per-head rmsnorm on a slice of the fused QKV buffer (Qwen3 q/k-norm on
// strided rows): batch b, head h -> row at buf + b*ld + off + h*hd.
__global__ void k_rmsnorm_rows_s(float *__restrict__ buf, int ld, int off,
                                 int heads, const float *__restrict__ w,
                                 int hd, float eps) {
    const int b = blockIdx.x / heads;
    const int h = blockIdx.x % heads;
    float *row = buf + (int64_t)b * ld + off + h * hd;
    float acc = 0.0f;
    for (int i = threadIdx.x; i < hd; i += blockDim.x) {
        const float v = row[i];
        acc += v * v;
    }
    acc = wave_reduce_sum(acc);  // blockDim == 64
    const float inv = rsqrtf(acc / hd + eps);
    for (int i = threadIdx.x; i < hd; i += blockDim.x)
        row[i] = row[i] * inv * w[i];
}

// ---------------------------------------------- fused residual+rmsnorm(+q80)
// x[row] += partial[row] (ADD), then rmsnorm with w; output either f32 y
// (!QUANT) or the Q80 triple (QUANT). Fuses the reference's
// merge_add -> inv_rms -> rms_norm -> cast chain (llm.cpp:263-270) into one
// kernel: one extra pass saved per layer half, and the B=1 row gets a full
// 1024-thread workgroup with float4 traffic.
template <bool ADD, bool QUANT>
__global__ void k_add_rmsnorm(float *__restrict__ x,
                              const float *__restrict__ partial,
                              const float *__restrict__ w,
                              float *__restrict__ y,
                              int8_t *__restrict__ q,
                              float *__restrict__ s,
                              float *__restrict__ bs,
                              int n, float eps) {
    extern __shared__ float lds[];  // n floats: x staged once, re-read from LDS
    const int64_t base = (int64_t)blockIdx.x * n;
    float acc = 0.0f;
    for (int i = threadIdx.x * 4; i < n; i += blockDim.x * 4) {
        float4 v = *reinterpret_cast<const float4 *>(x + base + i);
        if (ADD) {
            const float4 p = *reinterpret_cast<const float4 *>(partial + base + i);
            v.x += p.x; v.y += p.y; v.z += p.z; v.w += p.w;
            *reinterpret_cast<float4 *>(x + base + i) = v;
        }
        *reinterpret_cast<float4 *>(lds + i) = v;
        acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    __shared__ float red[16];
    acc = wave_reduce_sum(acc);
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = acc;
    __syncthreads();
    const int nw = blockDim.x / WAVE;
    float total = 0.0f;
    for (int i = 0; i < nw; i++) total += red[i];
    const float inv = rsqrtf(total / n + eps);

    if (!QUANT) {
        for (int i = threadIdx.x * 4; i < n; i += blockDim.x * 4) {
            const float4 v = *reinterpret_cast<const float4 *>(lds + i);
            const float4 wv = *reinterpret_cast<const float4 *>(w + i);
            float4 o;
            o.x = v.x * inv * wv.x; o.y = v.y * inv * wv.y;
            o.z = v.z * inv * wv.z; o.w = v.w * inv * wv.w;
            *reinterpret_cast<float4 *>(y + base + i) = o;
        }
    } else {
        const int nb = n / QB;
        const int lane32 = threadIdx.x & 31;
        for (int blk = threadIdx.x / 32; blk < nb; blk += blockDim.x / 32) {
            const int i = blk * QB + lane32;
            const float v = lds[i] * inv * w[i];
            const float amax = group32_reduce_max(fabsf(v));
            const float d = amax / 127.0f;
            const float qinv = d > 0.0f ? 1.0f / d : 0.0f;
            const float qf = rintf(v * qinv);
            q[base + i] = (int8_t)qf;
            const float bsum = group32_reduce_sum(qf);
            if (lane32 == 0) {
                s[(int64_t)blockIdx.x * nb + blk] = d;
                bs[(int64_t)blockIdx.x * nb + blk] = bsum;
            }
        }
    }
}

// embedding row gather (replaces torch index_select inside the decode graph
// — the ATen gather kernel costs ~40 us/step there; reference OP_EMBEDDING,
// nn-cpu-ops.cpp:982-1008).
__global__ void k_embed_gather(const float *__restrict__ table,
                               const long *__restrict__ tokens,
                               float *__restrict__ x, int dim,
                               float *__restrict__ ssq) {
    const int b = blockIdx.y;
    const int64_t src = (int64_t)tokens[b] * dim;
    float local = 0.0f;
    for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 4; i < dim;
         i += gridDim.x * blockDim.x * 4) {
        const float4 v = *reinterpret_cast<const float4 *>(table + src + i);
        *reinterpret_cast<float4 *>(x + (int64_t)b * dim + i) = v;
        local += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    if (ssq != nullptr) {
        local = wave_reduce_sum(local);
        __shared__ float red[16];
        const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
        if (lane == 0) red[wid] = local;
        __syncthreads();
        if (threadIdx.x == 0) {
            float t = 0.0f;
            for (int i = 0; i < blockDim.x / WAVE; i++) t += red[i];
            atomicAdd(ssq + (b * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD, t);
        }
    }
}

// ------------------------------------------- wide norm from precomputed ssq
// The residual row's sum-of-squares arrives precomputed (GEMV RESID epilogue
// / embed / merge-add), so the rmsnorm becomes one wide pass with no
// in-kernel reduction: quantized (Q80 triple) or f32 output.
__global__ void k_norm_quant(const float *__restrict__ x,
                             const float *__restrict__ w,
                             const float *__restrict__ ssq,
                             int8_t *__restrict__ q,
                             float *__restrict__ s,
                             float *__restrict__ bs,
                             float *__restrict__ yout,
                             int n, float eps, int deferred) {
    const int b = blockIdx.y;
    // deferred: emit x*w with the inv_rms factored out of the scale — the
    // consuming GEMV applies inv from ssq (PRO==2); codes are identical
    // either way (q depends only on intra-block ratios)
    const float inv = deferred ? 1.0f : rsqrtf(ssq_total(ssq, b) / n + eps);
    const int nb = n / QB;
    const int gid = blockIdx.x * blockDim.x + threadIdx.x;
    const int blk = gid / 32;
    const int lane = threadIdx.x & 31;
    if (blk >= nb) return;
    const int i = blk * QB + lane;
    const float v = x[(int64_t)b * n + i] * inv * w[i];
    if (yout != nullptr) yout[(int64_t)b * n + i] = v;
    const float amax = group32_reduce_max(fabsf(v));
    const float dd = amax / 127.0f;
    const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
    const float qf = rintf(v * qinv);
    q[(int64_t)b * n + i] = (int8_t)qf;
    const float bsum = group32_reduce_sum(qf);
    if (lane == 0) {
        s[(int64_t)b * nb + blk] = dd;
        bs[(int64_t)b * nb + blk] = bsum;
    }
}

__global__ void k_norm_f32(const float *__restrict__ x,
                           const float *__restrict__ w,
                           const float *__restrict__ ssq,
                           float *__restrict__ y,
                           int n, float eps) {
    const int b = blockIdx.y;
    const float inv = rsqrtf(ssq_total(ssq, b) / n + eps);
    for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 4; i < n;
         i += gridDim.x * blockDim.x * 4) {
        const float4 v = *reinterpret_cast<const float4 *>(x + (int64_t)b * n + i);
        const float4 wv = *reinterpret_cast<const float4 *>(w + i);
        float4 o;
        o.x = v.x * inv * wv.x; o.y = v.y * inv * wv.y;
        o.z = v.z * inv * wv.z; o.w = v.w * inv * wv.w;
        *reinterpret_cast<float4 *>(y + (int64_t)b * n + i) = o;
    }
}

// x[b] += p[b] with per-row sum-of-squares accumulation (residual fold for
// paths that produce a separate partial: MoE weighted sum, f32 TP sync).
__global__ void k_add_ssq(float *__restrict__ x,
                          const float *__restrict__ p,
                          float *__restrict__ ssq, int n) {
    const int b = blockIdx.y;
    float local = 0.0f;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        const float v = x[(int64_t)b * n + i] + p[(int64_t)b * n + i];
        x[(int64_t)b * n + i] = v;
        local += v * v;
    }
    local = wave_reduce_sum(local);
    __shared__ float red[16];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.0f;
        for (int i = 0; i < blockDim.x / WAVE; i++) t += red[i];
        atomicAdd(ssq + (b * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD, t);
    }
}

// KV-cache dtype helpers (f16 KV is the default; torch extensions build
// with __HIP_NO_HALF_CONVERSIONS__, so conversions must be explicit)
__device__ __forceinline__ float kv_f(float v) { return v; }
__device__ __forceinline__ float kv_f(__half v) { return __half2float(v); }
template <typename T> __device__ __forceinline__ T kv_c(float v);
template <> __device__ __forceinline__ float kv_c<float>(float v) { return v; }
template <> __device__ __forceinline__ __half kv_c<__half>(float v) { return __float2half(v); }

// argmax pack: monotonic unsigned ordering of (float value, smallest index
// wins ties) for a single global atomicMax — on-device greedy sampling.
__device__ __forceinline__ unsigned long long argmax_pack(float v, int idx) {
    unsigned u = __float_as_uint(v);
    u = (u & 0x80000000u) ? ~u : (u | 0x80000000u);
    return ((unsigned long long)u << 32) | (unsigned)(0x7FFFFFFF - idx);
}

// stage 2 of the on-device greedy argmax: reduce the per-workgroup bests the
// logits GEMV wrote to scratch (no atomics anywhere — a single-slot atomicMax
// across 32k workgroups serializes on one cacheline and costs >1 ms).
__global__ void k_token_from_argmax(long *__restrict__ token,
                                    const unsigned long long *__restrict__ scratch,
                                    int count) {
    unsigned long long best = 0ull;
    for (int i = threadIdx.x; i < count; i += blockDim.x)
        best = max(best, scratch[i]);
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        best = max(best, (unsigned long long)__shfl_xor((long long)best, off, WAVE));
    __shared__ unsigned long long red[16];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = best;
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int i = 1; i < blockDim.x / WAVE; i++) best = max(best, red[i]);
        best = max(best, red[0]);
        token[0] = (long)(0x7FFFFFFF - (int)(best & 0xFFFFFFFFu));
    }
}

// ------------------------------------------------------------------ Q40 GEMV
// y[b, row] = sum_j w[row, j] * x[b, j]  with W in Q40 planes and x in Q80.
// One wave per output row; lane l streams block l, l+64, ... of the row.
// Weight layout: qs uint8 [d, n/2] (16B per block, byte j = elem j | elem
// j+16 << 4), scales f16 [d, n/32]. Per block:
//   true = sw*sx*(sum_i q_i*x_i - 8*sum_i x_i)  with q in 0..15
// so the nibble unpack is 2 VALU ops per 8 elems and sdot4 does the MAC
// (role of reference matmul_Q80_Q40_F32, nn-cpu-ops.cpp:231-449, and the
//  matmul-forward-q80-q40-f32.comp Vulkan shader).
__device__ __forceinline__ int q40_block_dot(const uint4 wq, const int4 x0,
                                             const int4 x1) {
    const uint32_t wv[4] = {wq.x, wq.y, wq.z, wq.w};
    const int32_t xv[8] = {x0.x, x0.y, x0.z, x0.w, x1.x, x1.y, x1.z, x1.w};
    int idot = 0;
    #pragma unroll
    for (int wi = 0; wi < 4; wi++) {
        idot = dot4((int)(wv[wi] & 0x0F0F0F0Fu), xv[wi], idot);            // elems 4wi..4wi+3
        idot = dot4((int)((wv[wi] >> 4) & 0x0F0F0F0Fu), xv[4 + wi], idot); // elems 16+4wi..
    }
    return idot;
}

// GEMV epilogue modes: fusing the ops that FOLLOW a matmul into its tail
// removes whole kernels from the per-layer chain (the reference runs each
// as its own op, llm.cpp:263-557).
#define EPI_NONE 0
#define EPI_PACK 4     // TP: plain dot + Q80 WIRE emit (codes then f16 scales,
                       // the all-gather payload) straight from the epilogue
#define EPI_RESID_Q 3  // EPI_RESID + deferred-scale Q80 emit of x*wnorm (one
                       // block per 16-wave wg; consumer applies inv_rms, PRO==2)
#define EPI_RESID 1  // x[b,row] += v; accumulate sum(x'^2) into 16-way-spread
                     //  slots -> the next norm becomes a single wide pass
#define EPI_ROPE 2   // llama rope: rows (2j,2j+1) are a rotation pair held by
                     //  one RPW=2 wave; q rotated into y, k rotated into the
                     //  KV cache, v copied into the cache

// ssq accumulators are spread over SSQ_SPREAD slots per batch row: a
// single-address atomicAdd from every workgroup serializes on one cacheline
// (measured: doubled the 4096-row GEMV).

// PRO=1 prologue: the GEMV consumes the f32 residual row directly, applying
// rmsnorm (from the precomputed ssq) and Q80-quantizing lane-locally —
// numerically identical to the separate norm+cast kernels, but zero extra
// launches. Each lane quantizes only the blocks it dots.
__device__ __forceinline__ void q80_quant_block(const float *__restrict__ xsrc,
                                                const float *__restrict__ wsrc,
                                                const float inv,
                                                int4 *lo, int4 *hi,
                                                float *scale, float *bsum) {
    float v[QB];
    #pragma unroll
    for (int t = 0; t < 8; t++) {
        const float4 xv = reinterpret_cast<const float4 *>(xsrc)[t];
        const float4 wv = reinterpret_cast<const float4 *>(wsrc)[t];
        v[4 * t] = xv.x * inv * wv.x;
        v[4 * t + 1] = xv.y * inv * wv.y;
        v[4 * t + 2] = xv.z * inv * wv.z;
        v[4 * t + 3] = xv.w * inv * wv.w;
    }
    float amax = 0.0f;
    #pragma unroll
    for (int i = 0; i < QB; i++) amax = fmaxf(amax, fabsf(v[i]));
    const float dd = amax / 127.0f;
    const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
    int words[8];
    float bs = 0.0f;
    #pragma unroll
    for (int t = 0; t < 8; t++) {
        int w = 0;
        #pragma unroll
        for (int e = 0; e < 4; e++) {
            const float qf = rintf(v[4 * t + e] * qinv);
            bs += qf;
            w |= ((int)qf & 0xFF) << (8 * e);
        }
        words[t] = w;
    }
    *lo = make_int4(words[0], words[1], words[2], words[3]);
    *hi = make_int4(words[4], words[5], words[6], words[7]);
    *scale = dd;
    *bsum = bs;
}

// RPW = rows per wave: processing 2 rows per wave doubles the independent
// 16B weight loads in flight per lane (the decode GEMV is HBM-latency
// limited at 1 row/wave). NB = batch columns.
template <int NB, int RPW, int EPI, int PRO>
__global__ void k_q40_gemv(const uint8_t *__restrict__ qs,
                           const __half *__restrict__ scales,
                           const int8_t *__restrict__ xq,
                           const float *__restrict__ xs,
                           const float *__restrict__ xbs,
                           float *__restrict__ y,
                           int d, int n,
                           unsigned long long *__restrict__ amax_scratch,
                           float *__restrict__ x_resid,
                           float *__restrict__ ssq,
                           const float *__restrict__ rope_cache,
                           const int *__restrict__ pos,
                           float *__restrict__ kc,
                           float *__restrict__ vc,
                           int q_dim0, int kv_dim0, int hd,
                           const float *__restrict__ xf,
                           const float *__restrict__ wnorm,
                           const float *__restrict__ ssq_in,
                           float eps, int kv_f16,
                           int8_t *__restrict__ oqq,
                           float *__restrict__ oqs,
                           float *__restrict__ oqbs) {
    const int wpb = blockDim.x / WAVE;
    const int wid = threadIdx.x / WAVE;
    const int row0 = (blockIdx.x * wpb + wid) * RPW;
    const int lane = threadIdx.x % WAVE;
    const int nb = n / QB;
    const int nbp = nb >> 1;
    // K-split (EPI_NONE, B=1 only): gridDim.y slices of the block-pair
    // range; the small-d down-projections underfill the chip at 1 wg per
    // 8 rows, so slicing K multiplies the resident workgroups
    const int nks = gridDim.y;
    const int j0p = (int)((int64_t)nbp * blockIdx.y / nks);
    const int j1p = (int)((int64_t)nbp * (blockIdx.y + 1) / nks);
    const bool wave_valid = row0 < d;
    const int rbase = wave_valid ? row0 : 0;
    [[maybe_unused]] __shared__ float svq[32];  // EPI_RESID_Q block staging

    const uint4 *wrow[RPW];
    const __half *srow[RPW];
    #pragma unroll
    for (int r = 0; r < RPW; r++) {
        const int row = min(rbase + r, d - 1);
        wrow[r] = reinterpret_cast<const uint4 *>(qs + (int64_t)row * (n >> 1));
        srow[r] = scales + (int64_t)row * nb;
    }

    float acc[RPW][NB];
    #pragma unroll
    for (int r = 0; r < RPW; r++)
        #pragma unroll
        for (int b = 0; b < NB; b++) acc[r][b] = 0.0f;

    float invb[NB];
    if (PRO == 1) {
        // PRO1 needs inv for the in-loop requantize; PRO2 only applies it
        // in the epilogue — computing it there keeps the ssq load off the
        // kernel's critical start (ahead of the first weight loads)
        #pragma unroll
        for (int b = 0; b < NB; b++)
            invb[b] = rsqrtf(ssq_total_wave(ssq_in, b, lane) / n + eps);
    }

    for (int jp = j0p + lane; jp < j1p; jp += WAVE) {
        const int j = jp << 1;
        uint4 wq0[RPW], wq1[RPW];
        float2 sw[RPW];
        #pragma unroll
        for (int r = 0; r < RPW; r++) {
            wq0[r] = wrow[r][j];
            wq1[r] = wrow[r][j + 1];
            sw[r] = __half22float2(*reinterpret_cast<const __half2 *>(srow[r] + j));
        }
        #pragma unroll
        for (int b = 0; b < NB; b++) {
            int4 x0, x1, x2, x3;
            float2 sx, bsum;
            if (PRO == 1) {
                q80_quant_block(xf + (int64_t)b * n + j * QB, wnorm + j * QB,
                                invb[b], &x0, &x1, &sx.x, &bsum.x);
                q80_quant_block(xf + (int64_t)b * n + (j + 1) * QB,
                                wnorm + (j + 1) * QB, invb[b], &x2, &x3,
                                &sx.y, &bsum.y);
            } else {
                const int4 *xrow = reinterpret_cast<const int4 *>(xq + (int64_t)b * n) + j * 2;
                x0 = xrow[0]; x1 = xrow[1]; x2 = xrow[2]; x3 = xrow[3];
                sx = *reinterpret_cast<const float2 *>(xs + (int64_t)b * nb + j);
                bsum = *reinterpret_cast<const float2 *>(xbs + (int64_t)b * nb + j);
            }
            #pragma unroll
            for (int r = 0; r < RPW; r++) {
                const int idot0 = q40_block_dot(wq0[r], x0, x1);
                const int idot1 = q40_block_dot(wq1[r], x2, x3);
                acc[r][b] = fmaf(sw[r].x * sx.x, (float)idot0 - 8.0f * bsum.x, acc[r][b]);
                acc[r][b] = fmaf(sw[r].y * sx.y, (float)idot1 - 8.0f * bsum.y, acc[r][b]);
            }
        }
    }
    if ((nb & 1) && lane == 0 && blockIdx.y == nks - 1) {  // odd trailing block
        const int j = nb - 1;
        #pragma unroll
        for (int r = 0; r < RPW; r++) {
            const uint4 wq = wrow[r][j];
            const float sw1 = __half2float(srow[r][j]);
            #pragma unroll
            for (int b = 0; b < NB; b++) {
                int4 xb0, xb1;
                float sx1, bs1;
                if (PRO == 1) {
                    q80_quant_block(xf + (int64_t)b * n + j * QB, wnorm + j * QB,
                                    invb[b], &xb0, &xb1, &sx1, &bs1);
                } else {
                    const int4 *xb = reinterpret_cast<const int4 *>(xq + (int64_t)b * n) + j * 2;
                    xb0 = xb[0]; xb1 = xb[1];
                    sx1 = xs[(int64_t)b * nb + j];
                    bs1 = xbs[(int64_t)b * nb + j];
                }
                const int idot = q40_block_dot(wq, xb0, xb1);
                acc[r][b] = fmaf(sw1 * sx1, (float)idot - 8.0f * bs1, acc[r][b]);
            }
        }
    }

    unsigned long long wave_best = 0ull;
    float ssq_local[NB];
    #pragma unroll
    for (int b = 0; b < NB; b++) ssq_local[b] = 0.0f;
    const int pos0 = (EPI == EPI_ROPE) ? pos[0] : 0;

    #pragma unroll
    for (int b = 0; b < NB; b++) {
        float v[RPW];
        if (PRO == 2)
            invb[b] = rsqrtf(ssq_total_wave(ssq_in, b, lane) / n + eps);
        #pragma unroll
        for (int r = 0; r < RPW; r++) {
            v[r] = wave_reduce_sum(acc[r][b]);
            if (PRO == 2) v[r] *= invb[b];  // deferred inv_rms (scalar/row)
        }
        if (lane != 0 || !wave_valid) continue;
        if (EPI == EPI_ROPE) {
            // RPW==2: rows (rbase, rbase+1) form one llama rotation pair
            const int pb = pos0 + b;
            if (rbase < q_dim0) {
                const int j = (rbase % hd) >> 1;
                const float cr = rope_cache[(int64_t)pb * hd + 2 * j];
                const float ci = rope_cache[(int64_t)pb * hd + 2 * j + 1];
                y[(int64_t)b * d + rbase] = v[0] * cr - v[1 % RPW] * ci;
                y[(int64_t)b * d + rbase + 1] = v[0] * ci + v[1 % RPW] * cr;
            } else if (rbase < q_dim0 + kv_dim0) {
                const int rk = rbase - q_dim0;
                const int j = (rk % hd) >> 1;
                const float cr = rope_cache[(int64_t)pb * hd + 2 * j];
                const float ci = rope_cache[(int64_t)pb * hd + 2 * j + 1];
                const float o0 = v[0] * cr - v[1 % RPW] * ci;
                const float o1 = v[0] * ci + v[1 % RPW] * cr;
                // runtime KV-dtype branch: lane 0 only, once per wave —
                // free next to the weight stream (f16 KV is the default)
                if (kv_f16) {
                    __half *kh = reinterpret_cast<__half *>(kc);
                    kh[(int64_t)pb * kv_dim0 + rk] = __float2half(o0);
                    kh[(int64_t)pb * kv_dim0 + rk + 1] = __float2half(o1);
                } else {
                    kc[(int64_t)pb * kv_dim0 + rk] = o0;
                    kc[(int64_t)pb * kv_dim0 + rk + 1] = o1;
                }
            } else {
                const int rv = rbase - q_dim0 - kv_dim0;
                if (kv_f16) {
                    __half *vh = reinterpret_cast<__half *>(vc);
                    vh[(int64_t)pb * kv_dim0 + rv] = __float2half(v[0]);
                    if (RPW == 2)
                        vh[(int64_t)pb * kv_dim0 + rv + 1] = __float2half(v[1 % RPW]);
                } else {
                    vc[(int64_t)pb * kv_dim0 + rv] = v[0];
                    if (RPW == 2) vc[(int64_t)pb * kv_dim0 + rv + 1] = v[1 % RPW];
                }
            }
        } else {
            #pragma unroll
            for (int r = 0; r < RPW; r++) {
                const int row = rbase + r;
                if (row >= d) continue;
                if (EPI == EPI_RESID) {
                    const float xv = x_resid[(int64_t)b * d + row] + v[r];
                    x_resid[(int64_t)b * d + row] = xv;
                    ssq_local[b] += xv * xv;
                } else if (EPI == EPI_RESID_Q) {
                    // residual fold + stage the weighted value for the
                    // wg-local Q80 block (rows of this wg = one block)
                    const float xv = x_resid[(int64_t)b * d + row] + v[r];
                    x_resid[(int64_t)b * d + row] = xv;
                    ssq_local[b] += xv * xv;
                    svq[wid * RPW + r] = xv * wnorm[row];
                } else if (EPI == EPI_PACK) {
                    svq[wid * RPW + r] = v[r];  // partial, packed below
                } else {
                    y[((int64_t)b + (int64_t)blockIdx.y * NB) * d + row] = v[r];
                    if (NB == 1) wave_best = max(wave_best, argmax_pack(v[r], row));
                }
            }
        }
    }
    if (EPI == EPI_NONE && NB == 1 && amax_scratch != nullptr) {
        // stage 1 of the greedy argmax: one packed best per workgroup
        __shared__ unsigned long long wb[16];
        if (lane == 0) wb[wid] = wave_best;
        __syncthreads();
        if (threadIdx.x == 0) {
            unsigned long long best = wb[0];
            for (int i = 1; i < wpb; i++) best = max(best, wb[i]);
            amax_scratch[blockIdx.x] = best;
        }
    }
    if constexpr (EPI == EPI_PACK) {
        // wire layout (all-gather payload, B=1): int8 codes [d] then f16
        // block scales [d/16 bytes] — what k_sync_quant_pack produces and
        // k_merge_add consumes (no blocksum on the wire)
        __syncthreads();  // publish svq
        if (threadIdx.x < 32 && (int)(blockIdx.x * 32 + threadIdx.x) < d) {
            const float vq = svq[threadIdx.x];
            const float amax = group32_reduce_max(fabsf(vq));
            const float dd = amax / 127.0f;
            const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
            uint8_t *buf = reinterpret_cast<uint8_t *>(oqq);
            buf[blockIdx.x * QB + threadIdx.x] =
                (uint8_t)(int8_t)rintf(vq * qinv);
            if (threadIdx.x == 0) {
                const __half h = __float2half(dd);
                const uint16_t u = *reinterpret_cast<const uint16_t *>(&h);
                buf[d + 2 * blockIdx.x] = (uint8_t)(u & 0xFF);
                buf[d + 2 * blockIdx.x + 1] = (uint8_t)(u >> 8);
            }
        }
    }
    if constexpr (EPI == EPI_RESID_Q) {
        __shared__ float sredq[16];
        if (lane == 0) sredq[wid] = wave_valid ? ssq_local[0] : 0.0f;
        __syncthreads();  // also publishes svq
        if (threadIdx.x == 0) {
            float t = 0.0f;
            for (int wv = 0; wv < wpb; wv++) t += sredq[wv];
            atomicAdd(ssq + (blockIdx.x & (SSQ_SPREAD - 1)) * SSQ_PAD, t);
        }
        if (threadIdx.x < 32 && (int)(blockIdx.x * 32 + threadIdx.x) < d) {
            const float vq = svq[threadIdx.x];
            const float amax = group32_reduce_max(fabsf(vq));
            const float dd = amax / 127.0f;
            const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
            const float qf = rintf(vq * qinv);
            oqq[blockIdx.x * QB + threadIdx.x] = (int8_t)qf;
            const float bsq = group32_reduce_sum(qf);
            if (threadIdx.x == 0) {
                oqs[blockIdx.x] = dd;
                oqbs[blockIdx.x] = bsq;
            }
        }
    }
    if (EPI == EPI_RESID) {
        __shared__ float sred[4][NB];
        if (lane == 0)
            #pragma unroll
            for (int b = 0; b < NB; b++) sred[wid][b] = wave_valid ? ssq_local[b] : 0.0f;
        __syncthreads();
        if (threadIdx.x < NB) {
            float t = 0.0f;
            for (int wv = 0; wv < wpb; wv++) t += sred[wv][threadIdx.x];
            atomicAdd(ssq + (threadIdx.x * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD, t);
        }
    }
}

// ------------------------------------------------- split-K flash decode
// Stage 1: grid (H0, B, S); split sp covers t = (sp*4+wave) + 4*S*i — the
// t-range is spread over S*4 waves so long contexts fill the chip.
// Per-split (m, l, o) goes to scratch; stage 2 combines the S partials.
template <int VEC, typename KVT, bool FUSE = false>
__global__ void k_attn_split(const float *__restrict__ q, int q_ld,
                             const KVT *__restrict__ kc,
                             const KVT *__restrict__ vc,
                             const int *__restrict__ pos,
                             int n_heads0, int kv_mul, int kv_dim0, float scale,
                             float *__restrict__ ml_scratch,
                             float *__restrict__ o_scratch,
                             int *__restrict__ counter,
                             float *__restrict__ y,
                             int8_t *__restrict__ zq,
                             float *__restrict__ zs,
                             float *__restrict__ zbs) {
    (void)counter; (void)y; (void)zq; (void)zs; (void)zbs;
    const int h0 = blockIdx.x;
    const int b = blockIdx.y;
    const int sp = blockIdx.z;
    const int S = gridDim.z;
    const int hd = VEC * WAVE;
    const int plen = pos[0] + b + 1;
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int kv_off = (h0 / kv_mul) * hd;

    // Each wave's 4 groups of 16 lanes score 4 timesteps per iteration:
    // the serial online-softmax chain advances once per 4 t (a 1-t-per-wave
    // loop measured 22 us at pos=1024 — latency-chained, not memory-bound).
    const int VEC16 = VEC * 4;  // q/k elems per lane within a 16-lane group
    const int lane16 = lane & 15;
    const int group = lane >> 4;

    float qreg[VEC16];
    #pragma unroll
    for (int v = 0; v < VEC16; v++)
        qreg[v] = q[(int64_t)b * q_ld + h0 * hd + lane16 * VEC16 + v] * scale;

    float m = -1e30f, l = 0.0f, o[VEC];
    #pragma unroll
    for (int v = 0; v < VEC; v++) o[v] = 0.0f;

    // 4 t-chunks (16 timesteps) per online-softmax round: the K dots of all
    // 4 chunks are independent, so their HBM loads issue together instead of
    // serializing behind the softmax chain (the 1-chunk-per-round loop was
    // latency-bound: ~13 us at pos 1024 for ~1.3 us of KV traffic)
    const int stride = 16 * S;
    for (int tb0 = (sp * 4 + wave) * 4; tb0 < plen; tb0 += 4 * stride) {
        float su[4];
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            const int tg = tb0 + u * stride + group;
            float partial = 0.0f;
            if (tg < plen) {
                const KVT *krow = kc + (int64_t)tg * kv_dim0 + kv_off + lane16 * VEC16;
                #pragma unroll
                for (int v = 0; v < VEC16; v++)
                    partial = fmaf(qreg[v], kv_f(krow[v]), partial);
            }
            su[u] = group16_reduce_sum(partial);
        }
        float s16[16];
        float mn = m;
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            #pragma unroll
            for (int gg = 0; gg < 4; gg++) {
                float v = __shfl(su[u], gg * 16, WAVE);
                if (tb0 + u * stride + gg >= plen) v = -1e30f;
                s16[4 * u + gg] = v;
                mn = fmaxf(mn, v);
            }
        }
        const float f = __expf(m - mn);
        float w16[16];
        float lsum = 0.0f;
        #pragma unroll
        for (int i = 0; i < 16; i++) {
            w16[i] = __expf(s16[i] - mn);
            lsum += w16[i];
        }
        l = l * f + lsum;
        #pragma unroll
        for (int v = 0; v < VEC; v++) o[v] *= f;
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            #pragma unroll
            for (int gg = 0; gg < 4; gg++) {
                const int t = tb0 + u * stride + gg;
                if (t >= plen) continue;
                const KVT *vrow = vc + (int64_t)t * kv_dim0 + kv_off + lane * VEC;
                #pragma unroll
                for (int v = 0; v < VEC; v++)
                    o[v] = fmaf(w16[4 * u + gg], kv_f(vrow[v]), o[v]);
            }
        }
        m = mn;
    }

    __shared__ float sm[4], sl[4];
    __shared__ float so[4][VEC * WAVE];
    if (lane == 0) { sm[wave] = m; sl[wave] = l; }
    __syncthreads();
    const float M = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    const float fw = __expf(m - M);
    #pragma unroll
    for (int v = 0; v < VEC; v++) so[wave][lane * VEC + v] = o[v] * fw;
    __syncthreads();
    const int64_t slot = ((int64_t)b * n_heads0 + h0) * S + sp;
    if (wave == 0) {
        const float L = sl[0] * __expf(sm[0] - M) + sl[1] * __expf(sm[1] - M)
                      + sl[2] * __expf(sm[2] - M) + sl[3] * __expf(sm[3] - M);
        if constexpr (FUSE) {
            // S==1 single-kernel path (short contexts): normalize and emit
            // the Q80 triple of this head directly — no scratch round-trip,
            // no combine launch (the 2-kernel pair costs ~11 us in-graph at
            // pos<256 where the actual KV read is microseconds)
            const float invL = 1.0f / L;
            float vv[VEC];
            #pragma unroll
            for (int v = 0; v < VEC; v++) {
                const int i = lane * VEC + v;
                vv[v] = (so[0][i] + so[1][i] + so[2][i] + so[3][i]) * invL;
            }
            if constexpr (VEC == 2) {
                // lane holds elems (2*lane, 2*lane+1): block = 16-lane group
                const float amax = group16_reduce_max(fmaxf(fabsf(vv[0]), fabsf(vv[1])));
                const float dd = amax / 127.0f;
                const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
                const float q0 = rintf(vv[0] * qinv), q1 = rintf(vv[1] * qinv);
                const int64_t obase = ((int64_t)b * n_heads0 + h0) * hd + lane * 2;
                zq[obase] = (int8_t)q0;
                zq[obase + 1] = (int8_t)q1;
                const float bsum = group16_reduce_sum(q0 + q1);
                if ((lane & 15) == 0) {
                    const int blk = (h0 * hd + lane * 2) / QB;
                    zs[(int64_t)b * (n_heads0 * hd / QB) + blk] = dd;
                    zbs[(int64_t)b * (n_heads0 * hd / QB) + blk] = bsum;
                }
            } else {
                const float amax = group32_reduce_max(fabsf(vv[0]));
                const float dd = amax / 127.0f;
                const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
                const float q0 = rintf(vv[0] * qinv);
                zq[((int64_t)b * n_heads0 + h0) * hd + lane] = (int8_t)q0;
                const float bsum = group32_reduce_sum(q0);
                if ((lane & 31) == 0) {
                    const int blk = (h0 * hd + lane) / QB;
                    zs[(int64_t)b * (n_heads0 * hd / QB) + blk] = dd;
                    zbs[(int64_t)b * (n_heads0 * hd / QB) + blk] = bsum;
                }
            }
        } else {
            if (lane == 0) {
                ml_scratch[slot * 2] = M;
                ml_scratch[slot * 2 + 1] = L;
            }
            #pragma unroll
            for (int v = 0; v < VEC; v++) {
                const int i = lane * VEC + v;
                o_scratch[slot * hd + i] = so[0][i] + so[1][i] + so[2][i] + so[3][i];
            }
        }
    }

}

// GQA-grouped split variant: one workgroup owns ALL kv_mul query heads of
// a kv head (wave w = query head kvh*kv_mul+w), so the K/V rows stream
// through one CU's L1 once instead of kv_mul times through different XCD
// L2s (4x HBM amplification for 8B's 32q/8kv at long context). Waves are
// independent heads — no cross-wave merge; each writes its split partial
// directly. The model scales S by kv_mul to keep the grid full.
template <int VEC, typename KVT>
__global__ void k_attn_split_gqa(const float *__restrict__ q, int q_ld,
                                 const KVT *__restrict__ kc,
                                 const KVT *__restrict__ vc,
                                 const int *__restrict__ pos,
                                 int n_heads0, int kv_mul, int kv_dim0,
                                 float scale,
                                 float *__restrict__ ml_scratch,
                                 float *__restrict__ o_scratch) {
    const int kvh = blockIdx.x;
    const int b = blockIdx.y;
    const int sp = blockIdx.z;
    const int S = gridDim.z;
    const int hd = VEC * WAVE;
    const int plen = pos[0] + b + 1;
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int h0 = kvh * kv_mul + wave;
    const int kv_off = kvh * hd;
    const int VEC16 = VEC * 4;
    const int lane16 = lane & 15;
    const int group = lane >> 4;

    float qreg[VEC16];
    #pragma unroll
    for (int v = 0; v < VEC16; v++)
        qreg[v] = q[(int64_t)b * q_ld + h0 * hd + lane16 * VEC16 + v] * scale;

    float m = -1e30f, l = 0.0f, o[VEC];
    #pragma unroll
    for (int v = 0; v < VEC; v++) o[v] = 0.0f;

    // 16 consecutive timesteps per round (4 group-chunks x 4 unrolled);
    // all waves walk the same rounds, so the wg's K/V reads coalesce in L1
    for (int tb0 = sp * 16; tb0 < plen; tb0 += 16 * S) {
        float su[4];
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            const int tg = tb0 + u * 4 + group;
            float partial = 0.0f;
            if (tg < plen) {
                const KVT *krow = kc + (int64_t)tg * kv_dim0 + kv_off + lane16 * VEC16;
                #pragma unroll
                for (int v = 0; v < VEC16; v++)
                    partial = fmaf(qreg[v], kv_f(krow[v]), partial);
            }
            su[u] = group16_reduce_sum(partial);
        }
        float s16[16];
        float mn = m;
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            #pragma unroll
            for (int gg = 0; gg < 4; gg++) {
                float v = __shfl(su[u], gg * 16, WAVE);
                if (tb0 + u * 4 + gg >= plen) v = -1e30f;
                s16[4 * u + gg] = v;
                mn = fmaxf(mn, v);
            }
        }
        const float f = __expf(m - mn);
        float w16[16];
        float lsum = 0.0f;
        #pragma unroll
        for (int i = 0; i < 16; i++) {
            w16[i] = __expf(s16[i] - mn);
            lsum += w16[i];
        }
        l = l * f + lsum;
        #pragma unroll
        for (int v = 0; v < VEC; v++) o[v] *= f;
        #pragma unroll
        for (int u = 0; u < 4; u++) {
            #pragma unroll
            for (int gg = 0; gg < 4; gg++) {
                const int t = tb0 + u * 4 + gg;
                if (t >= plen) continue;
                const KVT *vrow = vc + (int64_t)t * kv_dim0 + kv_off + lane * VEC;
                #pragma unroll
                for (int v = 0; v < VEC; v++)
                    o[v] = fmaf(w16[4 * u + gg], kv_f(vrow[v]), o[v]);
            }
        }
        m = mn;
    }

    const int64_t slot = ((int64_t)b * n_heads0 + h0) * S + sp;
    if (lane == 0) {
        ml_scratch[slot * 2] = m;
        ml_scratch[slot * 2 + 1] = l;
    }
    #pragma unroll
    for (int v = 0; v < VEC; v++)
        o_scratch[slot * hd + lane * VEC + v] = o[v];
}

// Stage 2: combine the S split partials. QUANT=true additionally emits the
// Q80 triple of the attention output directly (each workgroup owns a whole
// head = hd/32 quant blocks), eliminating the separate cast kernel the
// reference runs before the wo matmul. (A fused last-block-combine variant
// measured 2x SLOWER: per-workgroup threadfence + tail serialization.)
template <int VEC, bool QUANT>
__global__ void k_attn_combine(const float *__restrict__ ml_scratch,
                               const float *__restrict__ o_scratch,
                               float *__restrict__ y,
                               int8_t *__restrict__ zq,
                               float *__restrict__ zs,
                               float *__restrict__ zbs,
                               int n_heads0, int S) {
    const int h0 = blockIdx.x;
    const int b = blockIdx.y;
    const int hd = VEC * WAVE;  // blockDim.x == hd
    const int64_t base = ((int64_t)b * n_heads0 + h0) * S;
    float M = -1e30f;
    for (int sp = 0; sp < S; sp++)
        M = fmaxf(M, ml_scratch[(base + sp) * 2]);
    float L = 0.0f;
    for (int sp = 0; sp < S; sp++)
        L += ml_scratch[(base + sp) * 2 + 1] * __expf(ml_scratch[(base + sp) * 2] - M);
    const float invL = 1.0f / L;
    const int i = threadIdx.x;
    float acc = 0.0f;
    for (int sp = 0; sp < S; sp++)
        acc += o_scratch[(base + sp) * hd + i]
             * __expf(ml_scratch[(base + sp) * 2] - M);
    const float v = acc * invL;
    if (!QUANT) {
        y[((int64_t)b * n_heads0 + h0) * hd + i] = v;
    } else {
        const float amax = group32_reduce_max(fabsf(v));
        const float d = amax / 127.0f;
        const float qinv = d > 0.0f ? 1.0f / d : 0.0f;
        const float qf = rintf(v * qinv);
        zq[((int64_t)b * n_heads0 + h0) * hd + i] = (int8_t)qf;
        const float bsum = group32_reduce_sum(qf);
        if ((i & 31) == 0) {
            const int blk = (h0 * hd + i) / QB;
            zs[(int64_t)b * (n_heads0 * hd / QB) + blk] = d;
            zbs[(int64_t)b * (n_heads0 * hd / QB) + blk] = bsum;
        }
    }
}

// Grouped (MoE) variant: weights [n_experts, d, ...]; slot s uses expert
// expert_idx[s] and input row slot_batch[s]; y [S, d]
// (reference 3-D expert matmul with index indirection, nn-core.hpp:209-213).
// LPP = lanes per 2-row pair. The lane-tiled default (since round 2)
// picks LPP = smallest pow2 >= nbp so every lane stays busy at MoE shapes
// where nbp < 64 (Qwen3-30B w2: nbp=12 left 52 of 64 lanes idle -> the
// ~5x-off-stream round-1 grouped GEMV; see tools/moe_gemv_probe.hip).
// DLLAMA_MOE_V2=0 reverts to the whole-wave (LPP=64) layout.
template <int LPP, bool GATE = false>
__global__ void k_q40_gemv_grouped(const uint8_t *__restrict__ qs,
                                   const __half *__restrict__ scales,
                                   const int8_t *__restrict__ xq,
                                   const float *__restrict__ xs,
                                   const float *__restrict__ xbs,
                                   const int *__restrict__ expert_idx,
                                   float *__restrict__ y,
                                   int d, int n, int k_slots,
                                   const float *__restrict__ router,
                                   int n_experts, int topk,
                                   const float *__restrict__ ssq_in,
                                   float eps) {
    constexpr int NGRP = WAVE / LPP;
    const int wpb = blockDim.x / WAVE;
    const int full_lane = threadIdx.x % WAVE;
    const int grp = full_lane / LPP;
    const int row0 = ((blockIdx.x * wpb + threadIdx.x / WAVE) * NGRP + grp) * 2;
    const int slot = blockIdx.y;
    if (row0 >= d) return;
    const int lane = full_lane % LPP;
    const int nb = n / QB;
    const int nbp = nb >> 1;
    int e;
    if constexpr (GATE) {
        // in-kernel gate: each wave recomputes the deterministic top-k from
        // the router logits — removes the k_moe_gate launch from the chain
        int gi[16];
        float gw[16];
        const int br = slot / topk, want = slot - br * topk;
        moe_gate_wave(router + (int64_t)br * n_experts, n_experts, topk,
                      full_lane, gi, gw);
        e = 0;
        #pragma unroll
        for (int t = 0; t < 16; t++)
            if (t == want) e = gi[t];
    } else {
        e = expert_idx[slot];
    }
    const int b = slot / k_slots;
    const int row1 = min(row0 + 1, d - 1);
    const uint4 *wrow0 = reinterpret_cast<const uint4 *>(
        qs + ((int64_t)e * d + row0) * (n >> 1));
    const uint4 *wrow1 = reinterpret_cast<const uint4 *>(
        qs + ((int64_t)e * d + row1) * (n >> 1));
    const __half *srow0 = scales + ((int64_t)e * d + row0) * nb;
    const __half *srow1 = scales + ((int64_t)e * d + row1) * nb;
    float acc0 = 0.0f, acc1 = 0.0f;
    for (int jp = lane; jp < nbp; jp += LPP) {
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
    if ((nb & 1) && lane == 0) {  // odd trailing block
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
    if (ssq_in != nullptr) {
        // deferred-scale input (see EPI_RESID_Q): apply inv_rms per row
        const float inv = rsqrtf(
            ssq_total_wave(ssq_in, b, threadIdx.x % WAVE) / n + eps);
        acc0 *= inv;
        acc1 *= inv;
    }
    if (lane == 0) {
        y[(int64_t)slot * d + row0] = acc0;
        if (row0 + 1 < d) y[(int64_t)slot * d + row0 + 1] = acc1;
    }
}

// ------------------------------------------------- int8 MFMA prefill GEMM
// Batched (prefill) Q40xQ80 matmul on the matrix cores:
// C[b][m] = sum_j sw[m,j]*sx[b,j] * (x_i8[b, j*32..] . w_i4[m, j*32..])
// One v_mfma_i32_32x32x32_i8 per (32-row, 32-batch, 1-block) tile — K=32
// matches the quant block exactly, so per-block scales stay exact: the i32
// tile is descaled into f32 accumulators after every MFMA.
// Fragment layouts verified on-device (tools/mfma_probe.hip):
//   A[m][k]: lane=(m&31)|((k>>4)<<5), byte=k&15 ; B[k][n] mirrored on n;
//   C[m][n]: col=lane&31, row=(r&3)+8*(r>>2)+4*(lane>>5).
// Weights ride as the B operand so each lane needs only ONE weight-row
// scale per block; the 16 batch-row x-scales come from wave-private LDS.
typedef int v4i32_t __attribute__((ext_vector_type(4)));
typedef int v16i32_t __attribute__((ext_vector_type(16)));

__global__ void __launch_bounds__(256)
k_q40_gemm(const uint8_t *__restrict__ qs,
           const __half *__restrict__ scales,
           const int8_t *__restrict__ xq,
           const float *__restrict__ xs,
           float *__restrict__ y,
           float *__restrict__ part,
           int d, int n, int batch) {
    // gridDim.y = K-splits: a pure M decomposition leaves <1 workgroup/CU
    // for mid-size d (224 wg for the 28672-row W13) — K-split partials
    // restore occupancy, combined by k_gemm_reduce
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int mbase = (blockIdx.x * 4 + wave) * 32;  // 32 weight rows per wave
    if (mbase >= d) return;
    const int nb = n / QB;
    const int ksplit = gridDim.y;
    const int j0 = (int)((int64_t)nb * blockIdx.y / ksplit);
    const int j1 = (int)((int64_t)nb * (blockIdx.y + 1) / ksplit);
    const int khi = lane >> 5;       // 0: elems 0..15 (lo nibbles), 1: 16..31 (hi)
    const int bcol = lane & 31;      // x batch row for the A fragment
    const int mcol = lane & 31;      // weight row within the tile (B operand)
    const int mrow = min(mbase + mcol, d - 1);
    const uint4 *wrow = reinterpret_cast<const uint4 *>(qs + (int64_t)mrow * (n >> 1));
    const __half *srow = scales + (int64_t)mrow * nb;

    float facc[16];
    #pragma unroll
    for (int r = 0; r < 16; r++) facc[r] = 0.0f;

    // 2-block unroll: two independent load->MFMA->descale chains keep two
    // 16B weight loads in flight per lane (single-chain was latency-bound:
    // MfmaUtil 1.5%)
    auto extract = [&](const uint4 &wq, v4i32_t &b) {
        const uint32_t wv[4] = {wq.x, wq.y, wq.z, wq.w};
        #pragma unroll
        for (int t = 0; t < 4; t++) {
            uint32_t sx_ = khi ? ((wv[t] >> 4) & 0x0F0F0F0Fu)
                               : (wv[t] & 0x0F0F0F0Fu);
            sx_ ^= 0x08080808u;
            b[t] = (int)(sx_ | (((sx_ >> 3) & 0x01010101u) * 0xF0u));
        }
    };
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
        extract(wq0, b0);
        extract(wq1, b1);
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
        extract(wrow[j], b);
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
            if (ksplit == 1) {
                if (brow < batch)
                    y[(int64_t)brow * d + mbase + mcol] = facc[r];
            } else {
                part[(((int64_t)blockIdx.y * 32) + brow) * d + mbase + mcol] = facc[r];
            }
        }
    }
}

// The default prefill GEMM since round 2 (DLLAMA_GEMM_V2=0 reverts; see
// tools/gemm_v2_probe.hip for the standalone A/B harness). Differences vs
// the round-1 k_q40_gemm:
//   - activation fragments + x-scales staged in LDS per 8-block chunk,
//     loaded once per workgroup and shared by all 4 waves,
//   - weight uint4 tiles prefetched through a 4-deep register ring,
//   - descale as float2 pairs (adjacent C rows) with broadcast ds_reads.
// 71 VGPR + 16 AGPR (v1: 126+30) -> 5 waves/SIMD, no spills.
#define GEMM_V2_CHUNK 8

__global__ void __launch_bounds__(256)
k_q40_gemm_v2(const uint8_t *__restrict__ qs,
              const __half *__restrict__ scales,
              const int8_t *__restrict__ xq,
              const float *__restrict__ xs,
              float *__restrict__ y,
              float *__restrict__ part,
              int d, int n, int batch) {
    __shared__ int8_t lds_a[2][GEMM_V2_CHUNK][32][QB];
    __shared__ float lds_s[2][GEMM_V2_CHUNK][32];
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int tid = threadIdx.x;
    const int mbase = (blockIdx.x * 4 + wave) * 32;
    const int nb = n / QB;
    const int ksplit = gridDim.y;
    const int j0 = (int)((int64_t)nb * blockIdx.y / ksplit);
    const int j1 = (int)((int64_t)nb * (blockIdx.y + 1) / ksplit);
    const int khi = lane >> 5;
    const int mcol = lane & 31;
    const int mrow = min(mbase + mcol, d - 1);
    const uint4 *wrow = reinterpret_cast<const uint4 *>(qs + (int64_t)mrow * (n >> 1));
    const __half *srow = scales + (int64_t)mrow * nb;
    const bool live = mbase < d;

    auto stage = [&](int buf, int jc) {
        const int nblk = min(GEMM_V2_CHUNK, j1 - jc);
        for (int u = tid; u < nblk * 32; u += 256) {
            const int b = u & 31;
            const int blk = u >> 5;
            *reinterpret_cast<uint4 *>(&lds_a[buf][blk][b][0]) =
                *reinterpret_cast<const uint4 *>(xq + (int64_t)b * n + (jc + blk) * QB);
            *reinterpret_cast<uint4 *>(&lds_a[buf][blk][b][16]) =
                *reinterpret_cast<const uint4 *>(xq + (int64_t)b * n + (jc + blk) * QB + 16);
            lds_s[buf][blk][b] = xs[(int64_t)b * nb + (jc + blk)];
        }
    };
    auto extract = [&](const uint4 &wq, v4i32_t &b) {
        const uint32_t wv[4] = {wq.x, wq.y, wq.z, wq.w};
        #pragma unroll
        for (int t = 0; t < 4; t++) {
            uint32_t s = khi ? ((wv[t] >> 4) & 0x0F0F0F0Fu) : (wv[t] & 0x0F0F0F0Fu);
            s ^= 0x08080808u;
            b[t] = (int)(s | (((s >> 3) & 0x01010101u) * 0xF0u));
        }
    };

    float2 facc[8];
    #pragma unroll
    for (int r = 0; r < 8; r++) facc[r] = make_float2(0.0f, 0.0f);

    stage(0, j0);
    // weight pipeline as NAMED scalars (cur/next): an indexed prefetch ring
    // array gets allocated to scratch (80 B/lane round-trip in the hot
    // loop — final code-object metadata shows it even when -Rpass remarks
    // claim zero spill); two named uint4s cannot spill, and the compiler's
    // vmcnt counting still overlaps the next load with the current MFMA
    uint4 wq_cur = {};
    float sw_cur = 0.0f;
    if (live && j0 < j1) {
        wq_cur = wrow[j0];
        sw_cur = __half2float(srow[j0]);
    }
    __syncthreads();

    int buf = 0;
    for (int jc = j0; jc < j1; jc += GEMM_V2_CHUNK, buf ^= 1) {
        const int nblk = min(GEMM_V2_CHUNK, j1 - jc);
        if (jc + GEMM_V2_CHUNK < j1) stage(buf ^ 1, jc + GEMM_V2_CHUNK);
        if (live) {
            #pragma unroll
            for (int jj = 0; jj < GEMM_V2_CHUNK; jj++) {
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
                extract(wq, b);
                v16i32_t iacc = {};
                iacc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a, b, iacc, 0, 0, 0);
                #pragma unroll
                for (int r2 = 0; r2 < 8; r2++) {
                    const int brow = ((2 * r2) & 3) + 8 * (r2 >> 1) + 4 * khi;
                    const float2 sx2 = *reinterpret_cast<const float2 *>(
                        &lds_s[buf][jj][brow]);
                    facc[r2].x = fmaf((float)iacc[2 * r2], sw * sx2.x, facc[r2].x);
                    facc[r2].y = fmaf((float)iacc[2 * r2 + 1], sw * sx2.y, facc[r2].y);
                }
            }
        }
        __syncthreads();
    }

    if (live && mbase + mcol < d) {
        #pragma unroll
        for (int r2 = 0; r2 < 8; r2++) {
            const int brow = ((2 * r2) & 3) + 8 * (r2 >> 1) + 4 * khi;
            if (ksplit == 1) {
                if (brow < batch)
                    y[(int64_t)brow * d + mbase + mcol] = facc[r2].x;
                if (brow + 1 < batch)
                    y[(int64_t)(brow + 1) * d + mbase + mcol] = facc[r2].y;
            } else {
                part[(((int64_t)blockIdx.y * 32) + brow) * d + mbase + mcol] = facc[r2].x;
                part[(((int64_t)blockIdx.y * 32) + brow + 1) * d + mbase + mcol] = facc[r2].y;
            }
        }
    }
}

__global__ void k_gemm_reduce(const float *__restrict__ part,
                              float *__restrict__ y,
                              int d, int batch, int ksplit) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < (int64_t)batch * d; i += (int64_t)gridDim.x * blockDim.x) {
        const int b = (int)(i / d);
        const int m = (int)(i % d);
        float acc = 0.0f;
        for (int k = 0; k < ksplit; k++)
            acc += part[(((int64_t)k * 32) + b) * d + m];
        y[(int64_t)b * d + m] = acc;
    }
}

// router GEMV with the FFN rmsnorm fused: logits[b][e] =
// inv_rms(b) * sum_i gate[e,i] * x[b,i] * wnorm[i] — the MoE deferred-quant
// path never materializes t_norm (reference computes rms_norm then the gate
// matmul as separate ops, llm.cpp:430-450)
__global__ void k_router_gemv_norm(const float *__restrict__ gate,
                                   const float *__restrict__ x,
                                   const float *__restrict__ wnorm,
                                   const float *__restrict__ ssq,
                                   float *__restrict__ logits,
                                   int n_experts, int dim, float eps) {
    const int e = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
    const int b = blockIdx.y;
    if (e >= n_experts) return;
    const int lane = threadIdx.x % WAVE;
    const float inv = rsqrtf(ssq_total_wave(ssq, b, lane) / dim + eps);
    const float4 *g4 = reinterpret_cast<const float4 *>(gate + (int64_t)e * dim);
    const float4 *x4 = reinterpret_cast<const float4 *>(x + (int64_t)b * dim);
    const float4 *w4 = reinterpret_cast<const float4 *>(wnorm);
    float acc = 0.0f;
    for (int i = lane; i < dim / 4; i += WAVE) {
        const float4 gv = g4[i];
        const float4 xv = x4[i];
        const float4 wv = w4[i];
        acc += gv.x * xv.x * wv.x + gv.y * xv.y * wv.y
             + gv.z * xv.z * wv.z + gv.w * xv.w * wv.w;
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) logits[(int64_t)b * n_experts + e] = acc * inv;
}

// f32 router GEMV: logits[b][e] = gate[e,:] . t[b,:]
__global__ void k_router_gemv(const float *__restrict__ gate,
                              const float *__restrict__ t,
                              float *__restrict__ logits,
                              int n_experts, int dim) {
    const int e = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
    const int b = blockIdx.y;
    if (e >= n_experts) return;
    const int lane = threadIdx.x % WAVE;
    const float4 *g4 = reinterpret_cast<const float4 *>(gate + (int64_t)e * dim);
    const float4 *t4 = reinterpret_cast<const float4 *>(t + (int64_t)b * dim);
    float acc = 0.0f;
    for (int i = lane; i < dim / 4; i += WAVE) {
        const float4 gv = g4[i];
        const float4 tv = t4[i];
        acc += gv.x * tv.x + gv.y * tv.y + gv.z * tv.z + gv.w * tv.w;
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) logits[(int64_t)b * n_experts + e] = acc;
}

// MoE router: softmax over n_experts logits, top-k (first-index ties),
// normalized weights + int32 expert ids (reference OP_SOFTMAX + OP_MOE_GATE,
// nn-cpu-ops.cpp:1443-1492). One wave per batch row; n_experts <= 1024.
// rank-select top-k (replaces the iterative wave-argmax version): expert
// e's output slot is its rank = #{j: p_j > p_e} + #{j < e: p_j == p_e}
// (identical ordering and smallest-index tie-break as the reference's
// insertion sort, nn-cpu-ops.cpp:900-916). All-pairs comparison is pure
// parallel VALU over LDS — the old 8 serial rounds of 6-step wave argmax
// were shuffle-latency-bound (~6 us wall for one wave; measured
// tools/fuse_shape_probe G-vs-F).
__global__ void k_moe_gate(const float *__restrict__ logits,
                           int *__restrict__ idx,
                           float *__restrict__ wts,
                           int n_experts, int topk) {
    const int b = blockIdx.x;
    const float *lrow = logits + (int64_t)b * n_experts;
    __shared__ float p[1024];
    __shared__ float red[16];
    __shared__ float topsum;
    const int t = threadIdx.x;           // blockDim == 128
    const int wid = t / WAVE, lane = t % WAVE;
    float m = -1e30f;
    for (int i = t; i < n_experts; i += blockDim.x) {
        const float v = lrow[i];
        p[i] = v;
        m = fmaxf(m, v);
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off, WAVE));
    if (lane == 0) red[wid] = m;
    __syncthreads();
    m = fmaxf(red[0], red[1]);
    // softmax numerator only: the full-softmax denominator cancels in the
    // top-k normalization (w_t = p_t / sum_topk p)
    for (int i = t; i < n_experts; i += blockDim.x)
        p[i] = __expf(p[i] - m);
    if (t == 0) topsum = 0.0f;
    __syncthreads();
    for (int i = t; i < n_experts; i += blockDim.x) {
        const float mine = p[i];
        int rank = 0;
        for (int j = 0; j < n_experts; j++) {
            const float o = p[j];
            rank += (o > mine) || (o == mine && j < i);
        }
        if (rank < topk) {
            idx[(int64_t)b * topk + rank] = i;
            wts[(int64_t)b * topk + rank] = mine;  // normalized below
            atomicAdd(&topsum, mine);
        }
    }
    __syncthreads();
    if (t < topk)
        wts[(int64_t)b * topk + t] /= topsum;
}

// weighted sum of expert outputs + residual fold + ssq (reference OP_SCALE +
// OP_MERGE_SUM + merge_add fused): x[b] += sum_s wts[b,s] * y[b*k+s].
// GATE=true recomputes the expert weights from the router logits in-kernel
// (same deterministic moe_gate_wave as the grouped GEMVs).
template <bool GATE = false>
__global__ void k_scale_merge_add(float *__restrict__ x,
                                  const float *__restrict__ y,
                                  const float *__restrict__ wts,
                                  float *__restrict__ ssq,
                                  int n, int topk, int n_experts) {
    const int b = blockIdx.y;
    float gw[16];
    if constexpr (GATE) {
        int gi[16];
        moe_gate_wave(wts + (int64_t)b * n_experts, n_experts, topk,
                      threadIdx.x % WAVE, gi, gw);
    }
    float local = 0.0f;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        float acc = x[(int64_t)b * n + i];
        if constexpr (GATE) {
            #pragma unroll
            for (int s = 0; s < 16; s++)
                if (s < topk)
                    acc = fmaf(gw[s], y[((int64_t)b * topk + s) * n + i], acc);
        } else {
            for (int s = 0; s < topk; s++)
                acc = fmaf(wts[(int64_t)b * topk + s],
                           y[((int64_t)b * topk + s) * n + i], acc);
        }
        x[(int64_t)b * n + i] = acc;
        local += acc * acc;
    }
    local = wave_reduce_sum(local);
    __shared__ float red[16];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.0f;
        for (int i = 0; i < blockDim.x / WAVE; i++) t += red[i];
        atomicAdd(ssq + (b * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD, t);
    }
}

// TP variant of the above: weighted expert sum into the partial buffer
// (no residual fold, no ssq — those happen after the Q80 sync merge-add;
// replaces the eager torch.sum fallback so TP MoE decode stays
// graph-capturable with zero ATen in the step)
template <bool GATE = false>
__global__ void k_scale_merge(float *__restrict__ partial,
                              const float *__restrict__ y,
                              const float *__restrict__ wts,
                              int n, int topk, int n_experts) {
    const int b = blockIdx.y;
    float gw[16];
    if constexpr (GATE) {
        int gi[16];
        moe_gate_wave(wts + (int64_t)b * n_experts, n_experts, topk,
                      threadIdx.x % WAVE, gi, gw);
    }
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        float acc = 0.0f;
        if constexpr (GATE) {
            #pragma unroll
            for (int s = 0; s < 16; s++)
                if (s < topk)
                    acc = fmaf(gw[s], y[((int64_t)b * topk + s) * n + i], acc);
        } else {
            for (int s = 0; s < topk; s++)
                acc = fmaf(wts[(int64_t)b * topk + s],
                           y[((int64_t)b * topk + s) * n + i], acc);
        }
        partial[(int64_t)b * n + i] = acc;
    }
}

// TP logits assembly: the row-split wcls puts rank w's slice at global vocab
// offset w*vocab0, so full logits for batch row b are the concatenation of
// gather[w, b, :] over w (reference gathers the same slices to root,
// nn-network.cpp:568-600 with onlyFromWorkerToRoot). One strided copy kernel
// instead of a per-token ATen permute+reshape.
__global__ void k_logits_concat(float *__restrict__ dst,
                                const float *__restrict__ src,
                                int nb, int vocab0, int world) {
    const int b = blockIdx.y;
    const int64_t total = (int64_t)world * vocab0;
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const int w = i / vocab0, j = i - (int64_t)w * vocab0;
        dst[(int64_t)b * total + i] =
            src[((int64_t)w * nb + b) * vocab0 + j];
    }
}

// generic argmax stage 1 over a flat f32 array (the TP greedy path: argmax
// of the gathered full logits). Each block writes its packed best to
// scratch[blockIdx]; k_token_from_argmax reduces scratch.
__global__ void k_argmax_stage1(const float *__restrict__ x, int64_t n,
                                unsigned long long *__restrict__ scratch) {
    unsigned long long best = 0ull;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        best = max(best, argmax_pack(x[i], (int)i));
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        best = max(best, (unsigned long long)__shfl_xor((long long)best, off, WAVE));
    __shared__ unsigned long long red[16];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = best;
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int i = 1; i < blockDim.x / WAVE; i++) best = max(best, red[i]);
        scratch[blockIdx.x] = best;
    }
}

// MoE deferred-quant epilogue: weighted expert sum + residual fold + ssq +
// DEFERRED Q80 emit of x*wnorm (the next matmul's input; consumer applies
// inv_rms). Each 256-thread wg owns 256 consecutive elements = 8 wg-local
// Q80 blocks, so the quantization is barrier-free per 32-lane group.
__global__ void k_scale_merge_add_q(float *__restrict__ x,
                                    const float *__restrict__ y,
                                    const float *__restrict__ wts,
                                    float *__restrict__ ssq,
                                    const float *__restrict__ wnorm,
                                    int8_t *__restrict__ oq,
                                    float *__restrict__ os,
                                    float *__restrict__ obs,
                                    int n, int topk) {
    const int b = blockIdx.y;
    const int i = blockIdx.x * 256 + threadIdx.x;
    float acc = x[(int64_t)b * n + i];
    for (int s = 0; s < topk; s++)
        acc = fmaf(wts[(int64_t)b * topk + s],
                   y[((int64_t)b * topk + s) * n + i], acc);
    x[(int64_t)b * n + i] = acc;
    // ssq: per-wg reduce then one spread-slot atomic
    float local = acc * acc;
    local = wave_reduce_sum(local);
    __shared__ float red[4];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
        atomicAdd(ssq + (b * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD,
                  red[0] + red[1] + red[2] + red[3]);
    }
    // deferred quant of x*wnorm: block = this thread's 32-lane group
    const float v = acc * wnorm[i];
    const float amax = group32_reduce_max(fabsf(v));
    const float dd = amax / 127.0f;
    const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
    const float qf = rintf(v * qinv);
    oq[(int64_t)b * n + i] = (int8_t)qf;
    const float bsum = group32_reduce_sum(qf);
    if ((threadIdx.x & 31) == 0) {
        const int blk = i / QB;
        os[(int64_t)b * (n / QB) + blk] = dd;
        obs[(int64_t)b * (n / QB) + blk] = bsum;
    }
}

// ------------------------------------------------------------------ rope
// style 0 = llama interleaved pairs (reference ropeLlama_F32,
// nn-cpu-ops.cpp:843-863), style 1 = falcon/neox half-rotated
// (ropeFalcon_F32, :865-885). cache [seq, hd/2, 2] = (cos, sin).
// x [B, dim0]; row b uses position pos[0]+b.
template <int STYLE>
__global__ void k_rope(float *__restrict__ x,
                       const float *__restrict__ cache,
                       const int *__restrict__ pos,
                       int dim0, int hd) {
    const int b = blockIdx.y;
    const int p = pos[0] + b;
    const int half = hd >> 1;
    float *row = x + (int64_t)b * dim0;
    const float *pc = cache + (int64_t)p * hd;  // hd floats = hd/2 pairs
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < dim0 / 2;
         i += gridDim.x * blockDim.x) {
        const int head = i / half;
        const int j = i % half;  // cache pair index
        int i0, i1;
        if (STYLE == 0) {          // pair (2j, 2j+1) within each head
            i0 = head * hd + 2 * j;
            i1 = i0 + 1;
        } else {                   // pair (j, j+hd/2) within each head
            i0 = head * hd + j;
            i1 = i0 + half;
        }
        const float cr = pc[2 * j];
        const float ci = pc[2 * j + 1];
        const float v0 = row[i0];
        const float v1 = row[i1];
        row[i0] = v0 * cr - v1 * ci;
        row[i1] = v0 * ci + v1 * cr;
    }
}

// ------------------------------------------- fused rope(q,k) + kv append
// Operates directly on the fused QKV GEMV output buffer ([B, ld] with
// q | k | v packed per row): rotates q in place, rotates k and writes it to
// the cache row pos+b, and copies v to the cache — one launch replacing
// rope(q), rope(k), kv_append (reference runs 4 separate ops here,
// llm.cpp:300-330).
template <int STYLE, typename KVT>
__global__ void k_rope_kv(float *__restrict__ qkv, int ld,
                          int q_dim0, int kv_dim0,
                          const float *__restrict__ cache,
                          const int *__restrict__ pos,
                          KVT *__restrict__ kc,
                          KVT *__restrict__ vc,
                          int hd) {
    const int b = blockIdx.y;
    const int p = pos[0] + b;
    const int half = hd >> 1;
    const float *pc = cache + (int64_t)p * hd;
    float *qrow = qkv + (int64_t)b * ld;
    float *krow = qrow + q_dim0;
    const float *vrow = krow + kv_dim0;
    const int qp = q_dim0 >> 1, kp = kv_dim0 >> 1, vq = kv_dim0 >> 2;
    const int total = qp + kp + vq;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += gridDim.x * blockDim.x) {
        if (i < qp + kp) {
            const bool is_q = i < qp;
            const int ii = is_q ? i : i - qp;
            const int head = ii / half;
            const int j = ii % half;
            int i0, i1;
            if (STYLE == 0) { i0 = head * hd + 2 * j; i1 = i0 + 1; }
            else            { i0 = head * hd + j;     i1 = i0 + half; }
            const float cr = pc[2 * j];
            const float ci = pc[2 * j + 1];
            float *src = is_q ? qrow : krow;
            const float v0 = src[i0];
            const float v1 = src[i1];
            const float o0 = v0 * cr - v1 * ci;
            const float o1 = v0 * ci + v1 * cr;
            if (is_q) { src[i0] = o0; src[i1] = o1; }
            else {
                KVT *dst = kc + (int64_t)p * kv_dim0;
                dst[i0] = kv_c<KVT>(o0); dst[i1] = kv_c<KVT>(o1);
            }
        } else {
            const int j = (i - qp - kp) * 4;
            const float4 vv = *reinterpret_cast<const float4 *>(vrow + j);
            KVT *dst = vc + (int64_t)p * kv_dim0 + j;
            dst[0] = kv_c<KVT>(vv.x); dst[1] = kv_c<KVT>(vv.y);
            dst[2] = kv_c<KVT>(vv.z); dst[3] = kv_c<KVT>(vv.w);
        }
    }
}

// ------------------------------------------------------------------ kv append
// copy k,v batch rows into the caches at row pos[0]+b (reference OP_SHIFT,
// nn-cpu-ops.cpp:1419-1441).
template <typename KVT>
__global__ void k_kv_append(const float *__restrict__ k,
                            const float *__restrict__ v,
                            KVT *__restrict__ kc,
                            KVT *__restrict__ vc,
                            const int *__restrict__ pos,
                            int kv_dim0) {
    const int b = blockIdx.y;
    const int p = pos[0] + b;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < kv_dim0;
         i += gridDim.x * blockDim.x) {
        kc[(int64_t)p * kv_dim0 + i] = kv_c<KVT>(k[(int64_t)b * kv_dim0 + i]);
        vc[(int64_t)p * kv_dim0 + i] = kv_c<KVT>(v[(int64_t)b * kv_dim0 + i]);
    }
}

// ------------------------------------------------------------------ swiglu (+q80)
// d = silu(a) * g, quantized straight to Q80 (reference OP_SILU + OP_MUL +
// OP_CAST fused; silu nn-cpu-ops.cpp:462-500).
template <bool GELU>
__global__ void k_swiglu_q80(const float *__restrict__ a,
                             const float *__restrict__ g,
                             int lda, int n,
                             int8_t *__restrict__ q,
                             float *__restrict__ s,
                             float *__restrict__ bs,
                             int n_blocks_total) {
    int gid = (blockIdx.x * blockDim.x + threadIdx.x) / 32;
    int lane = threadIdx.x & 31;
    if (gid >= n_blocks_total) return;
    const int nb = n / QB;
    const int r = gid / nb;
    const int i = (gid % nb) * QB + lane;
    float av = a[(int64_t)r * lda + i];
    float gv = g[(int64_t)r * lda + i];
    // silu (reference nn-cpu-ops.cpp:462) or tanh-approx gelu (:454)
    float act = GELU
        ? 0.5f * av * (1.0f + tanhf(0.797884560802865f * (av + 0.044715f * av * av * av)))
        : av / (1.0f + __expf(-av));
    float v = act * gv;
    float amax = group32_reduce_max(fabsf(v));
    float d = amax / 127.0f;
    float inv = d > 0.0f ? 1.0f / d : 0.0f;
    float qf = rintf(v * inv);
    q[gid * QB + lane] = (int8_t)qf;
    float bsum = group32_reduce_sum(qf);
    if (lane == 0) { s[gid] = d; bs[gid] = bsum; }
}

// fused W1|W3 GEMV + SwiGLU + Q80 quantize (decode B=1): one kernel replaces
// gemv(w13) + k_swiglu_q80 — per-kernel in-graph overhead is ~3-5 us, so at
// 32 layers/step the saved launch is worth ~150 us/token. The workgroup
// owns one Q80 output block (32 ff indices): wave w computes the W1 and W3
// rows of indices (32*bx+2w, +1) with 4 row streams in flight; the block's
// silu(a)*g values assemble in LDS and the first 32 lanes quantize.
// (reference runs matmul w1, matmul w3, silu, mul, cast as 5 ops,
// llm.cpp:430-448 / nn-cpu-ops.cpp:462-500.)
template <bool GELU>
__global__ __launch_bounds__(1024) void k_q40_gemv_swiglu(
        const uint8_t *__restrict__ qs, const __half *__restrict__ scales,
        const int8_t *__restrict__ xq, const float *__restrict__ xs,
        const float *__restrict__ xbs, int ff, int n,
        int8_t *__restrict__ oq, float *__restrict__ os,
        float *__restrict__ obs) {
    const int wave = threadIdx.x / WAVE;  // 16 waves, 2 ff indices each
    const int lane = threadIdx.x % WAVE;
    const int i0 = blockIdx.x * 32 + 2 * wave;
    const int nb = n / QB, nbp = nb >> 1;
    const int rows[4] = {i0, i0 + 1, ff + i0, ff + i0 + 1};
    const uint4 *wrow[4];
    const __half *srow[4];
    #pragma unroll
    for (int r = 0; r < 4; r++) {
        wrow[r] = reinterpret_cast<const uint4 *>(qs + (int64_t)rows[r] * (n >> 1));
        srow[r] = scales + (int64_t)rows[r] * nb;
    }
    float acc[4] = {0.0f, 0.0f, 0.0f, 0.0f};
    for (int jp = lane; jp < nbp; jp += WAVE) {
        const int j = jp << 1;
        const int4 *xr = reinterpret_cast<const int4 *>(xq) + j * 2;
        const int4 x0 = xr[0], x1 = xr[1], x2 = xr[2], x3 = xr[3];
        const float2 sx = *reinterpret_cast<const float2 *>(xs + j);
        const float2 bsum = *reinterpret_cast<const float2 *>(xbs + j);
        #pragma unroll
        for (int r = 0; r < 4; r++) {
            const uint4 w0 = wrow[r][j], w1 = wrow[r][j + 1];
            const float2 sw = __half22float2(*reinterpret_cast<const __half2 *>(srow[r] + j));
            acc[r] = fmaf(sw.x * sx.x, (float)q40_block_dot(w0, x0, x1) - 8.0f * bsum.x, acc[r]);
            acc[r] = fmaf(sw.y * sx.y, (float)q40_block_dot(w1, x2, x3) - 8.0f * bsum.y, acc[r]);
        }
    }
    if ((nb & 1) && lane == 0) {  // odd trailing block
        const int j = nb - 1;
        const int4 *xb = reinterpret_cast<const int4 *>(xq) + j * 2;
        const float sx1 = xs[j], bs1 = xbs[j];
        #pragma unroll
        for (int r = 0; r < 4; r++)
            acc[r] = fmaf(__half2float(srow[r][j]) * sx1,
                          (float)q40_block_dot(wrow[r][j], xb[0], xb[1]) - 8.0f * bs1,
                          acc[r]);
    }
    #pragma unroll
    for (int r = 0; r < 4; r++) acc[r] = wave_reduce_sum(acc[r]);
    __shared__ float sv[32];
    if (lane == 0) {
        #pragma unroll
        for (int t = 0; t < 2; t++) {
            const float av = acc[t], gv = acc[2 + t];
            const float act = GELU
                ? 0.5f * av * (1.0f + tanhf(0.797884560802865f * (av + 0.044715f * av * av * av)))
                : av / (1.0f + __expf(-av));
            sv[2 * wave + t] = act * gv;
        }
    }
    __syncthreads();
    if (threadIdx.x < 32) {
        const float v = sv[threadIdx.x];
        const float amax = group32_reduce_max(fabsf(v));
        const float dd = amax / 127.0f;
        const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
        const float qf = rintf(v * qinv);
        oq[(int64_t)blockIdx.x * QB + threadIdx.x] = (int8_t)qf;
        const float bsum = group32_reduce_sum(qf);
        if (threadIdx.x == 0) { os[blockIdx.x] = dd; obs[blockIdx.x] = bsum; }
    }
}

// grouped (MoE) variant of the fused W1|W3 GEMV + SwiGLU + Q80 emit, with
// the gate computed in-kernel from the router logits: for decode (B=1) this
// single launch replaces moe_gate + grouped w13 GEMV + swiglu_q80
// (reference runs repeat_z, gate matmul, softmax, moe_gate, grouped w1/w3,
// silu, mul, cast — llm.cpp:450-487). Workgroup = one Q80 block of one
// expert slot's output.
template <bool GELU>
__global__ __launch_bounds__(1024) void k_q40_gemv_grouped_swiglu(
        const uint8_t *__restrict__ qs, const __half *__restrict__ scales,
        const int8_t *__restrict__ xq, const float *__restrict__ xs,
        const float *__restrict__ xbs, int ff, int n,
        const float *__restrict__ router, int n_experts, int topk,
        int8_t *__restrict__ oq, float *__restrict__ os,
        float *__restrict__ obs) {
    const int wave = threadIdx.x / WAVE;  // 16 waves, 2 ff indices each
    const int lane = threadIdx.x % WAVE;
    const int slot = blockIdx.y;
    const int br = slot / topk, want = slot - br * topk;
    int gi[16];
    float gw[16];
    moe_gate_wave(router + (int64_t)br * n_experts, n_experts, topk, lane, gi, gw);
    int e = 0;
    #pragma unroll
    for (int t = 0; t < 16; t++)
        if (t == want) e = gi[t];
    const int i0 = blockIdx.x * 32 + 2 * wave;
    const int nb = n / QB, nbp = nb >> 1;
    const int rows[4] = {i0, i0 + 1, ff + i0, ff + i0 + 1};
    const uint4 *wrow[4];
    const __half *srow[4];
    #pragma unroll
    for (int r = 0; r < 4; r++) {
        const int64_t rbase = (int64_t)e * 2 * ff + rows[r];
        wrow[r] = reinterpret_cast<const uint4 *>(qs + rbase * (n >> 1));
        srow[r] = scales + rbase * nb;
    }
    const int8_t *xrow = xq + (int64_t)br * n;
    const float *xsr = xs + (int64_t)br * nb;
    const float *xbr = xbs + (int64_t)br * nb;
    float acc[4] = {0.0f, 0.0f, 0.0f, 0.0f};
    for (int jp = lane; jp < nbp; jp += WAVE) {
        const int j = jp << 1;
        const int4 *xr = reinterpret_cast<const int4 *>(xrow) + j * 2;
        const int4 x0 = xr[0], x1 = xr[1], x2 = xr[2], x3 = xr[3];
        const float2 sx = *reinterpret_cast<const float2 *>(xsr + j);
        const float2 bsum = *reinterpret_cast<const float2 *>(xbr + j);
        #pragma unroll
        for (int r = 0; r < 4; r++) {
            const uint4 w0 = wrow[r][j], w1 = wrow[r][j + 1];
            const float2 sw = __half22float2(*reinterpret_cast<const __half2 *>(srow[r] + j));
            acc[r] = fmaf(sw.x * sx.x, (float)q40_block_dot(w0, x0, x1) - 8.0f * bsum.x, acc[r]);
            acc[r] = fmaf(sw.y * sx.y, (float)q40_block_dot(w1, x2, x3) - 8.0f * bsum.y, acc[r]);
        }
    }
    if ((nb & 1) && lane == 0) {  // odd trailing block
        const int j = nb - 1;
        const int4 *xb = reinterpret_cast<const int4 *>(xrow) + j * 2;
        const float sx1 = xsr[j], bs1 = xbr[j];
        #pragma unroll
        for (int r = 0; r < 4; r++)
            acc[r] = fmaf(__half2float(srow[r][j]) * sx1,
                          (float)q40_block_dot(wrow[r][j], xb[0], xb[1]) - 8.0f * bs1,
                          acc[r]);
    }
    #pragma unroll
    for (int r = 0; r < 4; r++) acc[r] = wave_reduce_sum(acc[r]);
    __shared__ float sv[32];
    if (lane == 0) {
        #pragma unroll
        for (int t = 0; t < 2; t++) {
            const float av = acc[t], gv = acc[2 + t];
            const float act = GELU
                ? 0.5f * av * (1.0f + tanhf(0.797884560802865f * (av + 0.044715f * av * av * av)))
                : av / (1.0f + __expf(-av));
            sv[2 * wave + t] = act * gv;
        }
    }
    __syncthreads();
    if (threadIdx.x < 32) {
        const float v = sv[threadIdx.x];
        const float amax = group32_reduce_max(fabsf(v));
        const float dd = amax / 127.0f;
        const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
        const float qf = rintf(v * qinv);
        oq[((int64_t)slot * ff) + blockIdx.x * QB + threadIdx.x] = (int8_t)qf;
        const float bsum = group32_reduce_sum(qf);
        if (threadIdx.x == 0) {
            os[(int64_t)slot * (ff / QB) + blockIdx.x] = dd;
            obs[(int64_t)slot * (ff / QB) + blockIdx.x] = bsum;
        }
    }
}

// Qwen3 attention prologue in one launch: per-head rmsnorm of the q/k heads
// (reference OP_RMS_NORM nColumns mode, llm.cpp:178-187) + neox rope + KV
// cache write (reference rope + shift ops). One workgroup (1 wave) per head:
// q heads normalize+rotate in place, k heads normalize+rotate into the
// cache, v heads copy to the cache.
template <typename KVT>
__global__ void k_rope_kv_qknorm(float *__restrict__ qkv, int ld,
                                 int q_dim0, int kv_dim0,
                                 const float *__restrict__ cache,
                                 const int *__restrict__ pos,
                                 KVT *__restrict__ kc, KVT *__restrict__ vc,
                                 int hd,
                                 const float *__restrict__ wq,
                                 const float *__restrict__ wk, float eps) {
    const int b = blockIdx.y;
    const int p = pos[0] + b;
    const int h = blockIdx.x;
    const int qh = q_dim0 / hd, kvh = kv_dim0 / hd;
    const int half = hd >> 1;
    const float *pc = cache + (int64_t)p * hd;
    const int lane = threadIdx.x;  // blockDim == 64
    if (h >= qh + kvh) {
        // v head: plain copy into the cache
        const int hv = h - qh - kvh;
        const float *src = qkv + (int64_t)b * ld + q_dim0 + kv_dim0 + hv * hd;
        KVT *dst = vc + (int64_t)p * kv_dim0 + hv * hd;
        for (int i = lane; i < hd; i += WAVE) dst[i] = kv_c<KVT>(src[i]);
        return;
    }
    const bool is_q = h < qh;
    float *row = qkv + (int64_t)b * ld + (is_q ? h * hd
                                               : q_dim0 + (h - qh) * hd);
    const float *w = is_q ? wq : wk;
    float acc = 0.0f;
    for (int i = lane; i < hd; i += WAVE) {
        const float v = row[i];
        acc += v * v;
    }
    acc = wave_reduce_sum(acc);
    const float inv = rsqrtf(acc / hd + eps);
    for (int j = lane; j < half; j += WAVE) {
        const float v0 = row[j] * inv * w[j];
        const float v1 = row[j + half] * inv * w[j + half];
        const float cr = pc[2 * j], ci = pc[2 * j + 1];
        const float o0 = v0 * cr - v1 * ci;
        const float o1 = v0 * ci + v1 * cr;
        if (is_q) {
            row[j] = o0;
            row[j + half] = o1;
        } else {
            KVT *dst = kc + (int64_t)p * kv_dim0 + (h - qh) * hd;
            dst[j] = kv_c<KVT>(o0);
            dst[j + half] = kv_c<KVT>(o1);
        }
    }
}

__global__ void k_silu_mul(const float *__restrict__ a,
                           const float *__restrict__ g,
                           float *__restrict__ out,
                           int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        float av = a[i];
        out[i] = av / (1.0f + __expf(-av)) * g[i];
    }
}

// ------------------------------------------------------------------ q80 sync pack / merge-add
// wire layout per row: n int8 payload then n/32 f16 scales
// (role of cast-forward-f32-q80.comp; consumed after RCCL all-gather by
//  k_merge_add, the reference merge-add-forward-q80-f32.comp equivalent).
// fused quantize-into-wire: one pass replaces k_q80_quantize + k_sync_pack
// for the TP sync (the wire consumer k_merge_add needs no blocksum).
// Default since round 2 (DLLAMA_FUSED_SYNC=0 splits it again).
// EXPERIMENTAL round-2 path: dispatched only under DLLAMA_FUSED_SYNC=1.
__global__ void k_sync_quant_pack(const float *__restrict__ x,
                                  uint8_t *__restrict__ buf,
                                  int n, int n_blocks_total) {
    const int gid = (blockIdx.x * blockDim.x + threadIdx.x) / 32;
    const int lane = threadIdx.x & 31;
    if (gid >= n_blocks_total) return;
    const int nb = n / QB;
    const int r = gid / nb, jb = gid % nb;
    const int row_bytes = n + 2 * nb;
    const float v = x[(int64_t)gid * QB + lane];
    const float amax = group32_reduce_max(fabsf(v));
    const float d = amax / 127.0f;
    const float inv = d > 0.0f ? 1.0f / d : 0.0f;
    buf[(int64_t)r * row_bytes + jb * QB + lane] = (uint8_t)(int8_t)rintf(v * inv);
    if (lane == 0) {
        const __half h = __float2half(d);
        const uint16_t u = *reinterpret_cast<const uint16_t *>(&h);
        buf[(int64_t)r * row_bytes + n + 2 * jb] = (uint8_t)(u & 0xFF);
        buf[(int64_t)r * row_bytes + n + 2 * jb + 1] = (uint8_t)(u >> 8);
    }
}

__global__ void k_sync_pack(const int8_t *__restrict__ q,
                            const float *__restrict__ s,
                            uint8_t *__restrict__ buf,
                            int n, int rows) {
    const int nb = n / QB;
    const int row_bytes = n + 2 * nb;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < (int64_t)rows * n; i += (int64_t)gridDim.x * blockDim.x) {
        const int r = i / n, c = i % n;
        buf[(int64_t)r * row_bytes + c] = (uint8_t)q[i];
        if (c < nb) {
            const __half h = __float2half(s[(int64_t)r * nb + c]);
            const uint16_t u = *reinterpret_cast<const uint16_t *>(&h);
            buf[(int64_t)r * row_bytes + n + 2 * c] = (uint8_t)(u & 0xFF);
            buf[(int64_t)r * row_bytes + n + 2 * c + 1] = (uint8_t)(u >> 8);
        }
    }
}

// x[r, i] += sum_w dequant(bufs[w, r, i]) — the all-reduce completion
// (reference OP_MERGE_ADD, nn-cpu-ops.cpp:920-957).
__global__ void k_merge_add(float *__restrict__ x,
                            const uint8_t *__restrict__ bufs,
                            float *__restrict__ ssq,
                            int world, int n, int rows) {
    const int nb = n / QB;
    const int row_bytes = n + 2 * nb;
    const int r = blockIdx.y;
    float local = 0.0f;
    for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < n;
         c += gridDim.x * blockDim.x) {
        float acc = x[(int64_t)r * n + c];
        for (int w = 0; w < world; w++) {
            const uint8_t *row = bufs + ((int64_t)w * rows + r) * row_bytes;
            const int8_t qv = (int8_t)row[c];
            const uint16_t u = (uint16_t)row[n + 2 * (c / QB)]
                             | ((uint16_t)row[n + 2 * (c / QB) + 1] << 8);
            const __half h = *reinterpret_cast<const __half *>(&u);
            acc = fmaf((float)qv, __half2float(h), acc);
        }
        x[(int64_t)r * n + c] = acc;
        local += acc * acc;
    }
    if (ssq != nullptr) {
        local = wave_reduce_sum(local);
        __shared__ float red[16];
        const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
        if (lane == 0) red[wid] = local;
        __syncthreads();
        if (threadIdx.x == 0) {
            float t = 0.0f;
            for (int i = 0; i < blockDim.x / WAVE; i++) t += red[i];
            atomicAdd(ssq + (r * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD, t);
        }
    }
}

// TP MoE epilogue: expert-weighted sum of the grouped-GEMV outputs packed
// straight into the Q80 wire (the all-gather payload) — replaces
// scale_merge + sync_quant_pack on the TP critical path. One 256-thread wg
// per 256 elements (8 wire blocks), B=1 decode.
__global__ void k_scale_merge_pack(const float *__restrict__ y,
                                   const float *__restrict__ wts,
                                   uint8_t *__restrict__ buf,
                                   int n, int topk) {
    const int b = blockIdx.y;
    const int i = blockIdx.x * 256 + threadIdx.x;
    float acc = 0.0f;
    for (int s = 0; s < topk; s++)
        acc = fmaf(wts[(int64_t)b * topk + s],
                   y[((int64_t)b * topk + s) * n + i], acc);
    const float amax = group32_reduce_max(fabsf(acc));
    const float dd = amax / 127.0f;
    const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
    const int row_bytes = n + 2 * (n / QB);
    uint8_t *row = buf + (int64_t)b * row_bytes;
    row[i] = (uint8_t)(int8_t)rintf(acc * qinv);
    if ((threadIdx.x & 31) == 0) {
        const __half h = __float2half(dd);
        const uint16_t u = *reinterpret_cast<const uint16_t *>(&h);
        row[n + 2 * (i / QB)] = (uint8_t)(u & 0xFF);
        row[n + 2 * (i / QB) + 1] = (uint8_t)(u >> 8);
    }
}

// merge_add + DEFERRED Q80 emit of x*wnorm (the next matmul's input; the
// PRO==2 consumer applies inv_rms): completes the TP all-reduce AND
// replaces the following norm_quant launch. One 256-thread wg per 256
// elements = 8 wg-local quant blocks.
__global__ void k_merge_add_q(float *__restrict__ x,
                              const uint8_t *__restrict__ bufs,
                              float *__restrict__ ssq,
                              const float *__restrict__ wnorm,
                              int8_t *__restrict__ oq,
                              float *__restrict__ os,
                              float *__restrict__ obs,
                              int world, int n, int rows) {
    const int nb = n / QB;
    const int row_bytes = n + 2 * nb;
    const int r = blockIdx.y;
    const int c = blockIdx.x * 256 + threadIdx.x;
    float acc = x[(int64_t)r * n + c];
    for (int w = 0; w < world; w++) {
        const uint8_t *row = bufs + ((int64_t)w * rows + r) * row_bytes;
        const int8_t qv = (int8_t)row[c];
        const uint16_t u = (uint16_t)row[n + 2 * (c / QB)]
                         | ((uint16_t)row[n + 2 * (c / QB) + 1] << 8);
        const __half h = *reinterpret_cast<const __half *>(&u);
        acc = fmaf((float)qv, __half2float(h), acc);
    }
    x[(int64_t)r * n + c] = acc;
    float local = acc * acc;
    local = wave_reduce_sum(local);
    __shared__ float red[4];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = local;
    __syncthreads();
    if (threadIdx.x == 0)
        atomicAdd(ssq + (r * SSQ_SPREAD + (blockIdx.x & (SSQ_SPREAD - 1))) * SSQ_PAD,
                  red[0] + red[1] + red[2] + red[3]);
    const float v = acc * wnorm[c];
    const float amax = group32_reduce_max(fabsf(v));
    const float dd = amax / 127.0f;
    const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
    const float qf = rintf(v * qinv);
    oq[(int64_t)r * n + c] = (int8_t)qf;
    const float bsum = group32_reduce_sum(qf);
    if ((threadIdx.x & 31) == 0) {
        os[(int64_t)r * nb + c / QB] = dd;
        obs[(int64_t)r * nb + c / QB] = bsum;
    }
}

// K-split completion for the deferred down-projections: x += sum of the
// K-slices' partials, ssq, and the DEFERRED Q80 emit of x*wnorm — the
// same epilogue EPI_RESID_Q provides, but fed by a fully-filled K-split
// GEMV instead of a 1-wg-per-32-rows underfilled one. 256-elem wgs.
__global__ void k_add_ssq_q(float *__restrict__ x,
                            const float *__restrict__ parts,
                            float *__restrict__ ssq,
                            const float *__restrict__ wnorm,
                            int8_t *__restrict__ oq,
                            float *__restrict__ os,
                            float *__restrict__ obs,
                            int n, int nks) {
    const int i = blockIdx.x * 256 + threadIdx.x;
    float acc = x[i];
    for (int k = 0; k < nks; k++)
        acc += parts[(int64_t)k * n + i];
    x[i] = acc;
    float local = acc * acc;
    local = wave_reduce_sum(local);
    __shared__ float red[4];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) red[wid] = local;
    __syncthreads();
    if (threadIdx.x == 0)
        atomicAdd(ssq + (blockIdx.x & (SSQ_SPREAD - 1)) * SSQ_PAD,
                  red[0] + red[1] + red[2] + red[3]);
    const float v = acc * wnorm[i];
    const float amax = group32_reduce_max(fabsf(v));
    const float dd = amax / 127.0f;
    const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
    const float qf = rintf(v * qinv);
    oq[i] = (int8_t)qf;
    const float bsum = group32_reduce_sum(qf);
    if ((threadIdx.x & 31) == 0) {
        os[i / QB] = dd;
        obs[i / QB] = bsum;
    }
}

// x += y (residual merge for the f32/TP=1 path)
__global__ void k_add(float *__restrict__ x, const float *__restrict__ y, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        x[i] += y[i];
}

__global__ void k_inc(int *__restrict__ pos, int delta) {
    if (threadIdx.x == 0 && blockIdx.x == 0) pos[0] += delta;
}

#endif  // __HIPCC__

// ============================================================== host bindings

#define CHECK_CUDA(x) TORCH_CHECK((x).is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONT(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

static hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

void q80_quantize(torch::Tensor x, torch::Tensor q, torch::Tensor s, torch::Tensor bs) {
    CHECK_CUDA(x); CHECK_CONT(x);
    const int64_t blocks = x.numel() / QB;
    const int threads = 256;
    hipLaunchKernelGGL(k_q80_quantize, dim3(ceil_div(blocks * 32, threads)), dim3(threads),
                       0, cur_stream(), x.data_ptr<float>(), q.data_ptr<int8_t>(),
                       s.data_ptr<float>(), bs.data_ptr<float>(), (int)blocks);
}

void rmsnorm(torch::Tensor x, torch::Tensor w, torch::Tensor y, double eps) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int rows = x.numel() / n;
    hipLaunchKernelGGL(k_rmsnorm, dim3(rows), dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), w.data_ptr<float>(), y.data_ptr<float>(),
                       n, (float)eps);
}

void rmsnorm_q80(torch::Tensor x, torch::Tensor w, torch::Tensor q,
                 torch::Tensor s, torch::Tensor bs, double eps) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int rows = x.numel() / n;
    hipLaunchKernelGGL(k_rmsnorm_q80, dim3(rows), dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), w.data_ptr<float>(), q.data_ptr<int8_t>(),
                       s.data_ptr<float>(), bs.data_ptr<float>(), n, (float)eps);
}

void rmsnorm_rows(torch::Tensor x, torch::Tensor w, torch::Tensor y, double eps) {
    CHECK_CUDA(x);
    const int hd = x.size(-1);
    const int rows = x.numel() / hd;
    hipLaunchKernelGGL(k_rmsnorm_rows, dim3(rows), dim3(64), 0, cur_stream(),
                       x.data_ptr<float>(), w.data_ptr<float>(), y.data_ptr<float>(),
                       hd, (float)eps);
}

struct GemvEpi {
    unsigned long long *slot = nullptr;
    float *x_resid = nullptr;
    float *ssq = nullptr;
    const float *cache = nullptr;
    const int *pos = nullptr;
    float *kc = nullptr;  // f32 or f16 rows; kv_f16 says which
    float *vc = nullptr;
    int q_dim0 = 0, kv_dim0 = 0, hd = 0;
    int kv_f16 = 0;
    bool force_rpw2 = false;
    // PRO=1 (fused norm+quant prologue) inputs
    const float *xf = nullptr;
    const float *wnorm = nullptr;
    const float *ssq_in = nullptr;
    float eps = 0.0f;
    // EPI_RESID_Q deferred-quant outputs
    int8_t *oqq = nullptr;
    float *oqs = nullptr;
    float *oqbs = nullptr;
};

template <int EPI, int PRO>
static void gemv_launch(torch::Tensor &qs, torch::Tensor &scales, torch::Tensor &xq,
                        torch::Tensor &xs, torch::Tensor &xbs, float *y,
                        int64_t batch, const GemvEpi &e) {
    const int d = qs.size(0);
    const int n = qs.size(1) * 2;
    TORCH_CHECK(PRO == 1 || xq.size(-1) == n, "x width mismatch");
    TORCH_CHECK(n % QB == 0, "n must be a multiple of 32");
    // EPI_RESID_Q / EPI_PACK: 16 waves x RPW2 = 32 rows per wg = one block
    const int waves_per_block = (EPI == EPI_RESID_Q || EPI == EPI_PACK) ? 16 : 4;
    const dim3 block(waves_per_block * WAVE);
    auto launch = [&](auto nb_const, auto rpw_const) {
        constexpr int RPW = decltype(rpw_const)::value;
        const dim3 grid(ceil_div(d, waves_per_block * RPW));
        hipLaunchKernelGGL((k_q40_gemv<decltype(nb_const)::value, RPW, EPI, PRO>), grid,
                           block, 0, cur_stream(),
                           qs.data_ptr<uint8_t>(),
                           reinterpret_cast<const __half *>(scales.data_ptr<at::Half>()),
                           PRO == 1 ? nullptr : xq.data_ptr<int8_t>(),
                           PRO == 1 ? nullptr : xs.data_ptr<float>(),
                           PRO == 1 ? nullptr : xbs.data_ptr<float>(), y, d, n,
                           e.slot, e.x_resid, e.ssq,
                           e.cache, e.pos, e.kc, e.vc, e.q_dim0, e.kv_dim0, e.hd,
                           e.xf, e.wnorm, e.ssq_in, e.eps, e.kv_f16,
                           e.oqq, e.oqs, e.oqbs);
    };
    std::integral_constant<int, 1> r1;
    std::integral_constant<int, 2> r2;
    // constexpr: a plain `if` would still instantiate the other branch's
    // kernels (dead RPW combinations that additionally spill)
    if constexpr (EPI == EPI_ROPE) {
        // rotation pairs require RPW=2 at every batch size.
        // (`else` below matters: code after a returning `if constexpr` is
        // still instantiated, which kept dead spilled variants alive)
        switch (batch) {
            case 1: launch(std::integral_constant<int, 1>{}, r2); break;
            case 2: launch(std::integral_constant<int, 2>{}, r2); break;
            case 4: launch(std::integral_constant<int, 4>{}, r2); break;
            case 8:
            case 16:
            case 32:
                // PRO (fused norm prologue) is a batch<=4 decode path
                if constexpr (PRO) {
                    TORCH_CHECK(false, "fused-norm GEMV supports batch <= 4");
                } else {
                    if (batch == 8) launch(std::integral_constant<int, 8>{}, r2);
                    else if (batch == 16) launch(std::integral_constant<int, 16>{}, r2);
                    else launch(std::integral_constant<int, 32>{}, r2);
                }
                break;
            default: TORCH_CHECK(false, "unsupported batch ", batch);
        }
    } else {
        switch (batch) {
            case 1: {
                // RPW=2 halves the wave count for 2x per-lane loads in
                // flight; tunable crossover (DLLAMA_RPW1_MAX)
                static const int rpw1_max =
                    std::getenv("DLLAMA_RPW1_MAX") ? atoi(std::getenv("DLLAMA_RPW1_MAX")) : 0;
                if (e.force_rpw2 || (d >= 2048 && d > rpw1_max))
                    launch(std::integral_constant<int, 1>{}, r2);
                else
                    launch(std::integral_constant<int, 1>{}, r1);
                break;
            }
            case 2: launch(std::integral_constant<int, 2>{}, r2); break;
            case 4: launch(std::integral_constant<int, 4>{}, r2); break;
            case 8:
            case 16:
            case 32:
                if constexpr (PRO) {
                    TORCH_CHECK(false, "fused-norm GEMV supports batch <= 4");
                } else {
                    if (batch == 8) launch(std::integral_constant<int, 8>{}, r1);
                    else if (batch == 16) launch(std::integral_constant<int, 16>{}, r1);
                    else launch(std::integral_constant<int, 32>{}, r1);
                }
                break;
            default: TORCH_CHECK(false, "unsupported batch ", batch);
        }
    }
}

void q40_gemv(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
              torch::Tensor xs, torch::Tensor xbs, torch::Tensor y, int64_t batch,
              c10::optional<torch::Tensor> amax_slot = c10::nullopt,
              c10::optional<torch::Tensor> ssq_in = c10::nullopt,
              double eps = 0.0) {
    // ssq_in given: the xq triple is DEFERRED (scales exclude inv_rms, see
    // EPI_RESID_Q); the kernel applies inv = rsqrt(ssq/n + eps) per row
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    GemvEpi e;
    if (amax_slot.has_value())
        e.slot = reinterpret_cast<unsigned long long *>(amax_slot->data_ptr<int64_t>());
    if (ssq_in.has_value()) {
        e.ssq_in = ssq_in->data_ptr<float>();
        e.eps = (float)eps;
        gemv_launch<EPI_NONE, 2>(qs, scales, xq, xs, xbs, y.data_ptr<float>(), batch, e);
    } else {
        gemv_launch<EPI_NONE, 0>(qs, scales, xq, xs, xbs, y.data_ptr<float>(), batch, e);
    }
}

void q40_gemv_pack(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                   torch::Tensor xs, torch::Tensor xbs, torch::Tensor wire) {
    // TP down-projection: plain partial dot + Q80 WIRE emit in one launch
    // (replaces gemv-into-partial + sync_quant_pack). B=1 decode only.
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    const int d = qs.size(0);
    TORCH_CHECK(d % 32 == 0, "gemv_pack needs d % 32 == 0");
    GemvEpi e;
    e.oqq = reinterpret_cast<int8_t *>(wire.data_ptr<uint8_t>());
    e.force_rpw2 = true;
    gemv_launch<EPI_PACK, 0>(qs, scales, xq, xs, xbs, nullptr, 1, e);
}

void q40_gemv_ksplit(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                     torch::Tensor xs, torch::Tensor xbs, torch::Tensor part,
                     int64_t ks) {
    // B=1 K-split GEMV into part[ks, d] (see k_add_ssq_q for the merge)
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    const int d = qs.size(0);
    const int n = qs.size(1) * 2;
    TORCH_CHECK((int64_t)ks * d <= part.numel(), "part buffer too small");
    const dim3 grid(ceil_div(d, 8), ks);
    hipLaunchKernelGGL((k_q40_gemv<1, 2, EPI_NONE, 0>), grid, dim3(256), 0,
                       cur_stream(), qs.data_ptr<uint8_t>(),
                       reinterpret_cast<const __half *>(scales.data_ptr<at::Half>()),
                       xq.data_ptr<int8_t>(), xs.data_ptr<float>(),
                       xbs.data_ptr<float>(), part.data_ptr<float>(), d, n,
                       nullptr, nullptr, nullptr, nullptr, nullptr, nullptr,
                       nullptr, 0, 0, 0, nullptr, nullptr, nullptr, 0.0f, 0,
                       nullptr, nullptr, nullptr);
}

void add_ssq_q(torch::Tensor x, torch::Tensor parts, torch::Tensor ssq,
               torch::Tensor wnorm, torch::Tensor oq, torch::Tensor os,
               torch::Tensor obs, int64_t nks) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    TORCH_CHECK(n % 256 == 0, "add_ssq_q needs dim % 256 == 0");
    hipLaunchKernelGGL(k_add_ssq_q, dim3(n / 256), dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), parts.data_ptr<float>(),
                       ssq.data_ptr<float>(), wnorm.data_ptr<float>(),
                       oq.data_ptr<int8_t>(), os.data_ptr<float>(),
                       obs.data_ptr<float>(), n, (int)nks);
}

void q40_gemv_resid_q(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                      torch::Tensor xs, torch::Tensor xbs, torch::Tensor x,
                      torch::Tensor ssq, torch::Tensor wnorm, torch::Tensor oq,
                      torch::Tensor os, torch::Tensor obs) {
    // decode down-projection with fused residual + ssq + DEFERRED Q80 emit
    // of x*wnorm: replaces gemv_resid + norm_quant (the consumer runs with
    // ssq_in to apply inv_rms). B=1 only; 16-wave wgs (one block per wg).
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    const int d = qs.size(0);
    TORCH_CHECK(d % 32 == 0, "resid_q needs d % 32 == 0");
    GemvEpi e;
    e.x_resid = x.data_ptr<float>();
    e.ssq = ssq.data_ptr<float>();
    e.wnorm = wnorm.data_ptr<float>();
    e.oqq = oq.data_ptr<int8_t>();
    e.oqs = os.data_ptr<float>();
    e.oqbs = obs.data_ptr<float>();
    e.force_rpw2 = true;
    gemv_launch<EPI_RESID_Q, 0>(qs, scales, xq, xs, xbs, x.data_ptr<float>(), 1, e);
}

void q40_gemv_resid(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                    torch::Tensor xs, torch::Tensor xbs, torch::Tensor x,
                    torch::Tensor ssq, int64_t batch) {
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    TORCH_CHECK(batch <= 32, "RESID epilogue LDS sized for batch<=32");
    GemvEpi e;
    e.x_resid = x.data_ptr<float>();
    e.ssq = ssq.data_ptr<float>();
    gemv_launch<EPI_RESID, 0>(qs, scales, xq, xs, xbs, x.data_ptr<float>(), batch, e);
}

void q40_gemv_rope(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                   torch::Tensor xs, torch::Tensor xbs, torch::Tensor y,
                   int64_t batch, torch::Tensor cache, torch::Tensor pos,
                   torch::Tensor kc, torch::Tensor vc, int64_t q_dim0,
                   int64_t kv_dim0, int64_t head_dim,
                   c10::optional<torch::Tensor> ssq_in = c10::nullopt,
                   double eps = 0.0) {
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    GemvEpi e;
    e.cache = cache.data_ptr<float>();
    e.pos = pos.data_ptr<int>();
    e.kv_f16 = kc.scalar_type() == at::kHalf ? 1 : 0;
    e.kc = static_cast<float *>(kc.data_ptr());
    e.vc = static_cast<float *>(vc.data_ptr());
    e.q_dim0 = (int)q_dim0;
    e.kv_dim0 = (int)kv_dim0;
    e.hd = (int)head_dim;
    e.force_rpw2 = true;  // rotation pairs live in one RPW=2 wave
    if (ssq_in.has_value()) {  // deferred-scale input (see q40_gemv)
        e.ssq_in = ssq_in->data_ptr<float>();
        e.eps = (float)eps;
        gemv_launch<EPI_ROPE, 2>(qs, scales, xq, xs, xbs, y.data_ptr<float>(), batch, e);
    } else {
        gemv_launch<EPI_ROPE, 0>(qs, scales, xq, xs, xbs, y.data_ptr<float>(), batch, e);
    }
}

void q40_gemv_nq(torch::Tensor qs, torch::Tensor scales, torch::Tensor x,
                 torch::Tensor wnorm, torch::Tensor ssq, double eps,
                 torch::Tensor y, int64_t batch,
                 c10::optional<torch::Tensor> amax_slot = c10::nullopt) {
    // GEMV with fused rmsnorm+Q80-quantize prologue (decode batches)
    CHECK_CUDA(qs); CHECK_CONT(qs);
    TORCH_CHECK(batch <= 4, "PRO prologue is for small (decode) batches");
    GemvEpi e;
    e.xf = x.data_ptr<float>();
    e.wnorm = wnorm.data_ptr<float>();
    e.ssq_in = ssq.data_ptr<float>();
    e.eps = (float)eps;
    if (amax_slot.has_value())
        e.slot = reinterpret_cast<unsigned long long *>(amax_slot->data_ptr<int64_t>());
    gemv_launch<EPI_NONE, 1>(qs, scales, x, x, x, y.data_ptr<float>(), batch, e);
}

void q40_gemv_nq_rope(torch::Tensor qs, torch::Tensor scales, torch::Tensor x,
                      torch::Tensor wnorm, torch::Tensor ssq, double eps,
                      torch::Tensor y, int64_t batch, torch::Tensor cache,
                      torch::Tensor pos, torch::Tensor kc, torch::Tensor vc,
                      int64_t q_dim0, int64_t kv_dim0, int64_t head_dim) {
    CHECK_CUDA(qs); CHECK_CONT(qs);
    TORCH_CHECK(batch <= 4, "PRO prologue is for small (decode) batches");
    GemvEpi e;
    e.xf = x.data_ptr<float>();
    e.wnorm = wnorm.data_ptr<float>();
    e.ssq_in = ssq.data_ptr<float>();
    e.eps = (float)eps;
    e.cache = cache.data_ptr<float>();
    e.pos = pos.data_ptr<int>();
    e.kv_f16 = kc.scalar_type() == at::kHalf ? 1 : 0;
    e.kc = static_cast<float *>(kc.data_ptr());
    e.vc = static_cast<float *>(vc.data_ptr());
    e.q_dim0 = (int)q_dim0;
    e.kv_dim0 = (int)kv_dim0;
    e.hd = (int)head_dim;
    e.force_rpw2 = true;
    gemv_launch<EPI_ROPE, 1>(qs, scales, x, x, x, y.data_ptr<float>(), batch, e);
}

void q40_gemm(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
              torch::Tensor xs, torch::Tensor y, int64_t batch,
              c10::optional<torch::Tensor> part = c10::nullopt,
              int64_t variant = -1) {
    // int8-MFMA batched matmul (prefill path); xq/xs must have >=32 rows
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    const int d = qs.size(0);
    const int n = qs.size(1) * 2;
    TORCH_CHECK(xq.size(-1) == n, "x width mismatch");
    TORCH_CHECK(xq.size(0) >= 32, "gemm needs 32 padded batch rows");
    TORCH_CHECK(batch <= 32, "the MFMA fragment covers 32 batch rows; "
                "chunk larger prompts (rows past 32 would be dropped)");
    TORCH_CHECK(n % QB == 0, "n must be a multiple of 32");
    const int mtiles = ceil_div(d, 128);
    int ksplit = 1;
    if (part.has_value()) {
        ksplit = std::max(1, std::min(1024 / mtiles, 16));
        ksplit = std::min<int>(ksplit, n / QB);
        while (ksplit > 1 && (int64_t)ksplit * 32 * d > part->numel()) ksplit--;
    }
    float *pp = ksplit > 1 ? part->data_ptr<float>() : nullptr;
    // v2 (LDS-staged) is the default since round 2 (validated on hardware:
    // ~25% faster at all three prefill shapes); DLLAMA_GEMM_V2=0 reverts
    static const bool env_v2 = !std::getenv("DLLAMA_GEMM_V2")
        || atoi(std::getenv("DLLAMA_GEMM_V2")) != 0;
    const bool v2 = variant < 0 ? env_v2 : variant == 1;
    auto *kern = v2 ? k_q40_gemm_v2 : k_q40_gemm;
    hipLaunchKernelGGL(kern, dim3(mtiles, ksplit), dim3(256), 0,
                       cur_stream(), qs.data_ptr<uint8_t>(),
                       reinterpret_cast<const __half *>(scales.data_ptr<at::Half>()),
                       xq.data_ptr<int8_t>(), xs.data_ptr<float>(),
                       y.data_ptr<float>(), pp, d, n, (int)batch);
    if (ksplit > 1)
        hipLaunchKernelGGL(k_gemm_reduce, dim3(ceil_div((int64_t)batch * d, 256)),
                           dim3(256), 0, cur_stream(), pp, y.data_ptr<float>(),
                           d, (int)batch, ksplit);
}

int64_t q40_gemv_argmax_blocks(int64_t d) {
    // stage-1 argmax scratch entries for a batch-1 GEMV over d rows;
    // must mirror the RPW selection above
    const int rpw = d >= 2048 ? 2 : 1;
    return ceil_div(d, 4 * rpw);
}

void q40_gemv_grouped(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                      torch::Tensor xs, torch::Tensor xbs, torch::Tensor expert_idx,
                      torch::Tensor y, int64_t k_slots, int64_t variant = -1,
                      c10::optional<torch::Tensor> router = c10::nullopt,
                      int64_t topk = 0, int64_t n_slots_override = 0,
                      c10::optional<torch::Tensor> ssq_in = c10::nullopt,
                      double eps = 0.0) {
    // router given: expert ids come from the in-kernel gate over the router
    // logits [B, n_experts] (expert_idx is then ignored); n_slots_override
    // sets the slot count (it can no longer come from expert_idx.numel())
    CHECK_CUDA(qs); CHECK_CONT(qs);
    const int d = qs.size(1);
    const int n = qs.size(2) * 2;
    const int n_slots = n_slots_override > 0 ? (int)n_slots_override
                                             : (int)expert_idx.numel();
    const int waves_per_block = 4;
    // lane-tiled (v2) default since round 2 (validated: 1.4-1.8x at
    // Qwen3-30B shapes); DLLAMA_MOE_V2=0 reverts to the 64-lane layout
    static const bool env_v2 = !std::getenv("DLLAMA_MOE_V2")
        || atoi(std::getenv("DLLAMA_MOE_V2")) != 0;
    // v2: LPP = smallest power of two >= nbp (clamped [8, 64]) so every
    // lane has work; rows per wave scale up by 64/LPP
    int lpp = 64;
    if (variant < 0 ? env_v2 : variant == 1) {
        const int nbp = (n / QB) >> 1;
        lpp = 8;
        while (lpp < nbp && lpp < 64) lpp <<= 1;
    }
    const int rows_per_wg = waves_per_block * 2 * (WAVE / lpp);
    const dim3 grid(ceil_div(d, rows_per_wg), n_slots);
    const dim3 block(waves_per_block * WAVE);
    const bool gate = router.has_value();
    const float *rp = gate ? router->data_ptr<float>() : nullptr;
    const int ne = gate ? (int)router->size(-1) : 0;
    const float *sqp = ssq_in.has_value() ? ssq_in->data_ptr<float>() : nullptr;
    auto launch = [&](auto k) {
        hipLaunchKernelGGL(k, grid, block, 0, cur_stream(),
                           qs.data_ptr<uint8_t>(),
                           reinterpret_cast<const __half *>(scales.data_ptr<at::Half>()),
                           xq.data_ptr<int8_t>(), xs.data_ptr<float>(),
                           xbs.data_ptr<float>(), expert_idx.data_ptr<int>(),
                           y.data_ptr<float>(), d, n, (int)k_slots,
                           rp, ne, (int)topk, sqp, (float)eps);
    };
    if (gate) {
        switch (lpp) {
            case 8: launch(k_q40_gemv_grouped<8, true>); break;
            case 16: launch(k_q40_gemv_grouped<16, true>); break;
            case 32: launch(k_q40_gemv_grouped<32, true>); break;
            default: launch(k_q40_gemv_grouped<64, true>); break;
        }
    } else {
        switch (lpp) {
            case 8: launch(k_q40_gemv_grouped<8>); break;
            case 16: launch(k_q40_gemv_grouped<16>); break;
            case 32: launch(k_q40_gemv_grouped<32>); break;
            default: launch(k_q40_gemv_grouped<64>); break;
        }
    }
}

void rope(torch::Tensor x, torch::Tensor cache, torch::Tensor pos,
          int64_t head_dim, int64_t style) {
    CHECK_CUDA(x);
    const int B = x.size(0);
    const int dim0 = x.size(1);
    const dim3 grid(ceil_div(dim0 / 2, 256), B);
    if (style == 0)
        hipLaunchKernelGGL(k_rope<0>, grid, dim3(256), 0, cur_stream(),
                           x.data_ptr<float>(), cache.data_ptr<float>(),
                           pos.data_ptr<int>(), dim0, (int)head_dim);
    else
        hipLaunchKernelGGL(k_rope<1>, grid, dim3(256), 0, cur_stream(),
                           x.data_ptr<float>(), cache.data_ptr<float>(),
                           pos.data_ptr<int>(), dim0, (int)head_dim);
}

void kv_append(torch::Tensor k, torch::Tensor v, torch::Tensor kc,
               torch::Tensor vc, torch::Tensor pos) {
    CHECK_CUDA(k);
    const int B = k.size(0);
    const int kv_dim0 = k.size(1);
    const dim3 grid(ceil_div(kv_dim0, 256), B);
    if (kc.scalar_type() == at::kHalf)
        hipLaunchKernelGGL(k_kv_append<__half>, grid, dim3(256), 0, cur_stream(),
                           k.data_ptr<float>(), v.data_ptr<float>(),
                           reinterpret_cast<__half *>(kc.data_ptr<at::Half>()),
                           reinterpret_cast<__half *>(vc.data_ptr<at::Half>()),
                           pos.data_ptr<int>(), kv_dim0);
    else
        hipLaunchKernelGGL(k_kv_append<float>, grid, dim3(256), 0, cur_stream(),
                           k.data_ptr<float>(), v.data_ptr<float>(),
                           kc.data_ptr<float>(), vc.data_ptr<float>(),
                           pos.data_ptr<int>(), kv_dim0);
}

void attn(torch::Tensor q, int64_t q_ld, torch::Tensor kc, torch::Tensor vc,
          torch::Tensor y, torch::Tensor pos, int64_t batch, int64_t n_heads0,
          int64_t kv_mul, int64_t head_dim, int64_t splits,
          torch::Tensor ml_scratch, torch::Tensor o_scratch, torch::Tensor counter,
          c10::optional<torch::Tensor> zq = c10::nullopt,
          c10::optional<torch::Tensor> zs = c10::nullopt,
          c10::optional<torch::Tensor> zbs = c10::nullopt) {
    CHECK_CUDA(q);
    const int kv_dim0 = kc.size(1);
    const float scale = 1.0f / sqrtf((float)head_dim);
    const dim3 grid(n_heads0, batch, splits);
    const dim3 cgrid(n_heads0, batch);
    const bool quant = zq.has_value();
    const bool kv16 = kc.scalar_type() == at::kHalf;
    // GQA grouping experiment: one wg per (kv head, split) with kv_mul
    // waves sharing the K/V stream through L1. Measured ~20% SLOWER
    // end-to-end (448 vs 545 tok/s short-ctx 8B; MoE 366->255) — the L2
    // already absorbs the GQA re-reads and the grown combine fan-in costs
    // more than the saved traffic. Kept behind DLLAMA_GQA_ATTN=1.
    static const bool env_gqa =
        std::getenv("DLLAMA_GQA_ATTN") && atoi(std::getenv("DLLAMA_GQA_ATTN")) == 1;
    if (env_gqa && kv_mul > 1 && kv_mul <= 8 && n_heads0 % kv_mul == 0
        && splits > 1) {
        const dim3 ggrid(n_heads0 / kv_mul, batch, splits);
        const dim3 gblock(kv_mul * WAVE);
        auto grun = [&](auto vec_const, auto kvt, auto kcp, auto vcp) {
            constexpr int V = decltype(vec_const)::value;
            using KVT = decltype(kvt);
            hipLaunchKernelGGL((k_attn_split_gqa<V, KVT>), ggrid, gblock, 0,
                               cur_stream(), q.data_ptr<float>(), (int)q_ld,
                               kcp, vcp, pos.data_ptr<int>(), (int)n_heads0,
                               (int)kv_mul, kv_dim0, scale,
                               ml_scratch.data_ptr<float>(),
                               o_scratch.data_ptr<float>());
        };
        auto gcomb = [&](auto vec_const) {
            constexpr int V = decltype(vec_const)::value;
            if (quant)
                hipLaunchKernelGGL((k_attn_combine<V, true>), cgrid, dim3(V * WAVE),
                                   0, cur_stream(), ml_scratch.data_ptr<float>(),
                                   o_scratch.data_ptr<float>(), nullptr,
                                   zq->data_ptr<int8_t>(), zs->data_ptr<float>(),
                                   zbs->data_ptr<float>(), (int)n_heads0, (int)splits);
            else
                hipLaunchKernelGGL((k_attn_combine<V, false>), cgrid, dim3(V * WAVE),
                                   0, cur_stream(), ml_scratch.data_ptr<float>(),
                                   o_scratch.data_ptr<float>(), y.data_ptr<float>(),
                                   nullptr, nullptr, nullptr, (int)n_heads0, (int)splits);
        };
        auto gdisp = [&](auto vec_const) {
            if (kv16)
                grun(vec_const, __half{},
                     reinterpret_cast<const __half *>(kc.data_ptr<at::Half>()),
                     reinterpret_cast<const __half *>(vc.data_ptr<at::Half>()));
            else
                grun(vec_const, 0.0f, (const float *)kc.data_ptr<float>(),
                     (const float *)vc.data_ptr<float>());
            gcomb(vec_const);
        };
        if (head_dim == 128) gdisp(std::integral_constant<int, 2>{});
        else if (head_dim == 64) gdisp(std::integral_constant<int, 1>{});
        else TORCH_CHECK(false, "unsupported head_dim ", head_dim);
        return;
    }
    // splits==1 + quant output: single fused kernel (normalize + Q80 emit
    // in the split kernel), no combine launch
    if (splits == 1 && quant) {
        auto frun = [&](auto vec_const, auto kvt, auto kcp, auto vcp) {
            constexpr int V = decltype(vec_const)::value;
            using KVT = decltype(kvt);
            hipLaunchKernelGGL((k_attn_split<V, KVT, true>), grid, dim3(256), 0,
                               cur_stream(), q.data_ptr<float>(), (int)q_ld,
                               kcp, vcp, pos.data_ptr<int>(),
                               (int)n_heads0, (int)kv_mul, kv_dim0, scale,
                               ml_scratch.data_ptr<float>(),
                               o_scratch.data_ptr<float>(),
                               counter.data_ptr<int>(), nullptr,
                               zq->data_ptr<int8_t>(), zs->data_ptr<float>(),
                               zbs->data_ptr<float>());
        };
        auto fdisp = [&](auto vec_const) {
            if (kv16)
                frun(vec_const, __half{},
                     reinterpret_cast<const __half *>(kc.data_ptr<at::Half>()),
                     reinterpret_cast<const __half *>(vc.data_ptr<at::Half>()));
            else
                frun(vec_const, 0.0f, (const float *)kc.data_ptr<float>(),
                     (const float *)vc.data_ptr<float>());
        };
        if (head_dim == 128) fdisp(std::integral_constant<int, 2>{});
        else if (head_dim == 64) fdisp(std::integral_constant<int, 1>{});
        else TORCH_CHECK(false, "unsupported head_dim ", head_dim);
        return;
    }
    auto run = [&](auto vec_const) {
        constexpr int V = decltype(vec_const)::value;
        if (kv16)
            hipLaunchKernelGGL((k_attn_split<V, __half>), grid, dim3(256), 0,
                               cur_stream(), q.data_ptr<float>(), (int)q_ld,
                               reinterpret_cast<const __half *>(kc.data_ptr<at::Half>()),
                               reinterpret_cast<const __half *>(vc.data_ptr<at::Half>()),
                               pos.data_ptr<int>(),
                               (int)n_heads0, (int)kv_mul, kv_dim0, scale,
                               ml_scratch.data_ptr<float>(), o_scratch.data_ptr<float>(),
                               counter.data_ptr<int>(), nullptr, nullptr, nullptr, nullptr);
        else
            hipLaunchKernelGGL((k_attn_split<V, float>), grid, dim3(256), 0, cur_stream(),
                           q.data_ptr<float>(), (int)q_ld, kc.data_ptr<float>(),
                           vc.data_ptr<float>(), pos.data_ptr<int>(),
                           (int)n_heads0, (int)kv_mul, kv_dim0, scale,
                           ml_scratch.data_ptr<float>(), o_scratch.data_ptr<float>(),
                           counter.data_ptr<int>(), nullptr, nullptr, nullptr, nullptr);
        if (quant)
            hipLaunchKernelGGL((k_attn_combine<V, true>), cgrid, dim3(V * WAVE), 0,
                               cur_stream(), ml_scratch.data_ptr<float>(),
                               o_scratch.data_ptr<float>(), nullptr,
                               zq->data_ptr<int8_t>(), zs->data_ptr<float>(),
                               zbs->data_ptr<float>(), (int)n_heads0, (int)splits);
        else
            hipLaunchKernelGGL((k_attn_combine<V, false>), cgrid, dim3(V * WAVE), 0,
                               cur_stream(), ml_scratch.data_ptr<float>(),
                               o_scratch.data_ptr<float>(), y.data_ptr<float>(),
                               nullptr, nullptr, nullptr, (int)n_heads0, (int)splits);
    };
    if (head_dim == 128) run(std::integral_constant<int, 2>{});
    else if (head_dim == 64) run(std::integral_constant<int, 1>{});
    else TORCH_CHECK(false, "unsupported head_dim ", head_dim);
}

void embed_gather(torch::Tensor table, torch::Tensor tokens, torch::Tensor x,
                  int64_t batch, c10::optional<torch::Tensor> ssq = c10::nullopt) {
    CHECK_CUDA(table);
    const int dim = table.size(1);
    const dim3 grid(ceil_div(dim / 4, 256), batch);
    hipLaunchKernelGGL(k_embed_gather, grid, dim3(256), 0, cur_stream(),
                       table.data_ptr<float>(), (const long *)tokens.data_ptr<int64_t>(),
                       x.data_ptr<float>(), dim,
                       ssq.has_value() ? ssq->data_ptr<float>() : nullptr);
}

void norm_quant(torch::Tensor x, torch::Tensor w, torch::Tensor ssq,
                torch::Tensor q, torch::Tensor s, torch::Tensor bs,
                int64_t batch, double eps,
                c10::optional<torch::Tensor> yout = c10::nullopt,
                bool deferred = false) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const dim3 grid(ceil_div(n, 256), batch);
    hipLaunchKernelGGL(k_norm_quant, grid, dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), w.data_ptr<float>(), ssq.data_ptr<float>(),
                       q.data_ptr<int8_t>(), s.data_ptr<float>(), bs.data_ptr<float>(),
                       yout.has_value() ? yout->data_ptr<float>() : nullptr,
                       n, (float)eps, deferred ? 1 : 0);
}

void moe_gate(torch::Tensor logits, torch::Tensor idx, torch::Tensor wts,
              int64_t batch, int64_t topk) {
    CHECK_CUDA(logits);
    const int n_experts = logits.size(-1);
    TORCH_CHECK(n_experts <= 1024, "moe_gate supports <=1024 experts");
    hipLaunchKernelGGL(k_moe_gate, dim3(batch), dim3(128), 0, cur_stream(),
                       logits.data_ptr<float>(), idx.data_ptr<int>(),
                       wts.data_ptr<float>(), n_experts, (int)topk);
}

void scale_merge_add(torch::Tensor x, torch::Tensor y, torch::Tensor wts,
                     torch::Tensor ssq, int64_t batch, int64_t topk,
                     bool gate = false) {
    // gate=true: wts is the [B, n_experts] router-logits buffer and the
    // expert weights are recomputed in-kernel
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int ne = gate ? (int)wts.size(-1) : 0;
    auto launch = [&](auto k) {
        hipLaunchKernelGGL(k, dim3(ceil_div(n, 256), batch), dim3(256),
                           0, cur_stream(), x.data_ptr<float>(), y.data_ptr<float>(),
                           wts.data_ptr<float>(), ssq.data_ptr<float>(), n,
                           (int)topk, ne);
    };
    if (gate) launch(k_scale_merge_add<true>);
    else launch(k_scale_merge_add<false>);
}

void scale_merge(torch::Tensor partial, torch::Tensor y, torch::Tensor wts,
                 int64_t batch, int64_t topk, bool gate = false) {
    CHECK_CUDA(partial);
    const int n = partial.size(-1);
    const int ne = gate ? (int)wts.size(-1) : 0;
    auto launch = [&](auto k) {
        hipLaunchKernelGGL(k, dim3(ceil_div(n, 256), batch), dim3(256),
                           0, cur_stream(), partial.data_ptr<float>(),
                           y.data_ptr<float>(), wts.data_ptr<float>(), n,
                           (int)topk, ne);
    };
    if (gate) launch(k_scale_merge<true>);
    else launch(k_scale_merge<false>);
}

void logits_concat(torch::Tensor dst, torch::Tensor src, int64_t batch) {
    // src [world, nb, vocab0] (contiguous all-gather output) -> dst rows
    // 0..batch-1 of [.., world*vocab0]
    CHECK_CUDA(dst); CHECK_CONT(src);
    const int world = src.size(0);
    const int nb = src.size(1);
    const int vocab0 = src.size(2);
    const int64_t total = (int64_t)world * vocab0;
    hipLaunchKernelGGL(k_logits_concat,
                       dim3(std::min<int64_t>(ceil_div(total, 256), 2048), batch),
                       dim3(256), 0, cur_stream(), dst.data_ptr<float>(),
                       src.data_ptr<float>(), nb, vocab0, world);
}

void argmax_token(torch::Tensor token, torch::Tensor x, torch::Tensor scratch) {
    // two-stage argmax of flat f32 x into token[0] (TP greedy decode)
    CHECK_CUDA(x); CHECK_CONT(x);
    const int64_t n = x.numel();
    const int blocks = (int)std::min<int64_t>(scratch.numel(), ceil_div(n, 4096));
    auto *sp = reinterpret_cast<unsigned long long *>(scratch.data_ptr<int64_t>());
    hipLaunchKernelGGL(k_argmax_stage1, dim3(blocks), dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), n, sp);
    hipLaunchKernelGGL(k_token_from_argmax, dim3(1), dim3(1024), 0, cur_stream(),
                       (long *)token.data_ptr<int64_t>(), sp, blocks);
}

void scale_merge_add_q(torch::Tensor x, torch::Tensor y, torch::Tensor wts,
                       torch::Tensor ssq, torch::Tensor wnorm,
                       torch::Tensor oq, torch::Tensor os, torch::Tensor obs,
                       int64_t batch, int64_t topk) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    TORCH_CHECK(n % 256 == 0, "scale_merge_add_q needs dim % 256 == 0");
    hipLaunchKernelGGL(k_scale_merge_add_q, dim3(n / 256, batch), dim3(256),
                       0, cur_stream(), x.data_ptr<float>(), y.data_ptr<float>(),
                       wts.data_ptr<float>(), ssq.data_ptr<float>(),
                       wnorm.data_ptr<float>(), oq.data_ptr<int8_t>(),
                       os.data_ptr<float>(), obs.data_ptr<float>(), n, (int)topk);
}

void router_gemv_norm(torch::Tensor gate, torch::Tensor x, torch::Tensor wnorm,
                      torch::Tensor ssq, double eps, torch::Tensor logits,
                      int64_t batch) {
    CHECK_CUDA(gate);
    const int n_experts = gate.size(0);
    const int dim = gate.size(1);
    hipLaunchKernelGGL(k_router_gemv_norm, dim3(ceil_div(n_experts, 4), batch),
                       dim3(256), 0, cur_stream(), gate.data_ptr<float>(),
                       x.data_ptr<float>(), wnorm.data_ptr<float>(),
                       ssq.data_ptr<float>(), logits.data_ptr<float>(),
                       n_experts, dim, (float)eps);
}

void router_gemv(torch::Tensor gate, torch::Tensor t, torch::Tensor logits,
                 int64_t batch) {
    CHECK_CUDA(gate);
    const int n_experts = gate.size(0);
    const int dim = gate.size(1);
    hipLaunchKernelGGL(k_router_gemv, dim3(ceil_div(n_experts, 4), batch),
                       dim3(256), 0, cur_stream(), gate.data_ptr<float>(),
                       t.data_ptr<float>(), logits.data_ptr<float>(),
                       n_experts, dim);
}

void norm_f32(torch::Tensor x, torch::Tensor w, torch::Tensor ssq,
              torch::Tensor y, int64_t batch, double eps) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const dim3 grid(ceil_div(n / 4, 256), batch);
    hipLaunchKernelGGL(k_norm_f32, grid, dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), w.data_ptr<float>(), ssq.data_ptr<float>(),
                       y.data_ptr<float>(), n, (float)eps);
}

void add_ssq(torch::Tensor x, torch::Tensor p, torch::Tensor ssq, int64_t batch) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const dim3 grid(ceil_div(n, 1024), batch);
    hipLaunchKernelGGL(k_add_ssq, grid, dim3(256), 0, cur_stream(),
                       x.data_ptr<float>(), p.data_ptr<float>(),
                       ssq.data_ptr<float>(), n);
}

void swiglu_q80(torch::Tensor a, torch::Tensor g, int64_t lda, int64_t n,
                int64_t rows, torch::Tensor q, torch::Tensor s, torch::Tensor bs,
                bool gelu = false) {
    CHECK_CUDA(a);
    const int64_t blocks = rows * (n / QB);
    if (gelu)
        hipLaunchKernelGGL(k_swiglu_q80<true>, dim3(ceil_div(blocks * 32, 256)),
                           dim3(256), 0, cur_stream(), a.data_ptr<float>(),
                           g.data_ptr<float>(), (int)lda, (int)n,
                           q.data_ptr<int8_t>(), s.data_ptr<float>(),
                           bs.data_ptr<float>(), (int)blocks);
    else
        hipLaunchKernelGGL(k_swiglu_q80<false>, dim3(ceil_div(blocks * 32, 256)),
                           dim3(256), 0, cur_stream(), a.data_ptr<float>(),
                           g.data_ptr<float>(), (int)lda, (int)n,
                           q.data_ptr<int8_t>(), s.data_ptr<float>(),
                           bs.data_ptr<float>(), (int)blocks);
}

void rmsnorm_rows_s(torch::Tensor buf, int64_t ld, int64_t off, int64_t heads,
                    int64_t batch, torch::Tensor w, int64_t hd, double eps) {
    CHECK_CUDA(buf);
    hipLaunchKernelGGL(k_rmsnorm_rows_s, dim3(batch * heads), dim3(64), 0,
                       cur_stream(), buf.data_ptr<float>(), (int)ld, (int)off,
                       (int)heads, w.data_ptr<float>(), (int)hd, (float)eps);
}

void add_rmsnorm(torch::Tensor x, c10::optional<torch::Tensor> partial,
                 torch::Tensor w, torch::Tensor y, double eps) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int rows = x.numel() / n;
    const bool add = partial.has_value();
    const float *pp = add ? partial->data_ptr<float>() : nullptr;
    if (add)
        hipLaunchKernelGGL((k_add_rmsnorm<true, false>), dim3(rows), dim3(1024),
                           n * 4, cur_stream(), x.data_ptr<float>(), pp, w.data_ptr<float>(),
                           y.data_ptr<float>(), nullptr, nullptr, nullptr, n, (float)eps);
    else
        hipLaunchKernelGGL((k_add_rmsnorm<false, false>), dim3(rows), dim3(1024),
                           n * 4, cur_stream(), x.data_ptr<float>(), pp, w.data_ptr<float>(),
                           y.data_ptr<float>(), nullptr, nullptr, nullptr, n, (float)eps);
}

void add_rmsnorm_q80(torch::Tensor x, c10::optional<torch::Tensor> partial,
                     torch::Tensor w, torch::Tensor q, torch::Tensor s,
                     torch::Tensor bs, double eps) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int rows = x.numel() / n;
    const bool add = partial.has_value();
    const float *pp = add ? partial->data_ptr<float>() : nullptr;
    if (add)
        hipLaunchKernelGGL((k_add_rmsnorm<true, true>), dim3(rows), dim3(1024),
                           n * 4, cur_stream(), x.data_ptr<float>(), pp, w.data_ptr<float>(),
                           nullptr, q.data_ptr<int8_t>(), s.data_ptr<float>(),
                           bs.data_ptr<float>(), n, (float)eps);
    else
        hipLaunchKernelGGL((k_add_rmsnorm<false, true>), dim3(rows), dim3(1024),
                           n * 4, cur_stream(), x.data_ptr<float>(), pp, w.data_ptr<float>(),
                           nullptr, q.data_ptr<int8_t>(), s.data_ptr<float>(),
                           bs.data_ptr<float>(), n, (float)eps);
}

void rope_kv(torch::Tensor qkv, int64_t ld, int64_t q_dim0, int64_t kv_dim0,
             torch::Tensor cache, torch::Tensor pos, torch::Tensor kc,
             torch::Tensor vc, int64_t head_dim, int64_t style, int64_t batch) {
    CHECK_CUDA(qkv);
    const int total = (int)(q_dim0 / 2 + kv_dim0 / 2 + kv_dim0 / 4);
    const dim3 grid(ceil_div(total, 256), batch);
    auto run = [&](auto kvt, auto kcp, auto vcp) {
        using KVT = decltype(kvt);
        if (style == 0)
            hipLaunchKernelGGL((k_rope_kv<0, KVT>), grid, dim3(256), 0, cur_stream(),
                               qkv.data_ptr<float>(), (int)ld, (int)q_dim0, (int)kv_dim0,
                               cache.data_ptr<float>(), pos.data_ptr<int>(),
                               kcp, vcp, (int)head_dim);
        else
            hipLaunchKernelGGL((k_rope_kv<1, KVT>), grid, dim3(256), 0, cur_stream(),
                               qkv.data_ptr<float>(), (int)ld, (int)q_dim0, (int)kv_dim0,
                               cache.data_ptr<float>(), pos.data_ptr<int>(),
                               kcp, vcp, (int)head_dim);
    };
    if (kc.scalar_type() == at::kHalf)
        run(__half{}, reinterpret_cast<__half *>(kc.data_ptr<at::Half>()),
            reinterpret_cast<__half *>(vc.data_ptr<at::Half>()));
    else
        run(0.0f, kc.data_ptr<float>(), vc.data_ptr<float>());
}

void token_from_argmax(torch::Tensor token, torch::Tensor scratch, int64_t count) {
    hipLaunchKernelGGL(k_token_from_argmax, dim3(1), dim3(1024), 0, cur_stream(),
                       (long *)token.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned long long *>(scratch.data_ptr<int64_t>()),
                       (int)count);
}

void q40_gemv_swiglu(torch::Tensor qs, torch::Tensor scales, torch::Tensor xq,
                     torch::Tensor xs, torch::Tensor xbs, torch::Tensor oq,
                     torch::Tensor os, torch::Tensor obs, bool gelu = false) {
    // B=1 decode FFN: W1|W3 GEMV + SwiGLU + Q80 emit, one launch
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    const int d = qs.size(0);
    const int ff = d / 2;
    const int n = qs.size(1) * 2;
    TORCH_CHECK(ff % 32 == 0, "fused swiglu GEMV needs ff % 32 == 0");
    TORCH_CHECK(xq.size(-1) == n, "x width mismatch");
    const dim3 grid(ff / 32);
    const dim3 block(16 * WAVE);
    auto launch = [&](auto k) {
        hipLaunchKernelGGL(k, grid, block, 0, cur_stream(),
                           qs.data_ptr<uint8_t>(),
                           reinterpret_cast<const __half *>(scales.data_ptr<at::Half>()),
                           xq.data_ptr<int8_t>(), xs.data_ptr<float>(),
                           xbs.data_ptr<float>(), ff, n,
                           oq.data_ptr<int8_t>(), os.data_ptr<float>(),
                           obs.data_ptr<float>());
    };
    if (gelu) launch(k_q40_gemv_swiglu<true>);
    else launch(k_q40_gemv_swiglu<false>);
}

void q40_gemv_grouped_swiglu(torch::Tensor qs, torch::Tensor scales,
                             torch::Tensor xq, torch::Tensor xs,
                             torch::Tensor xbs, torch::Tensor oq,
                             torch::Tensor os, torch::Tensor obs,
                             int64_t n_slots, torch::Tensor router,
                             int64_t topk, bool gelu = false) {
    // decode MoE FFN up-projection: gate + grouped W1|W3 GEMV + SwiGLU +
    // Q80 emit in one launch. qs [E, 2*ff, n/2]; oq/os/obs are the
    // [n_slots, ff] Q80 triple.
    CHECK_CUDA(qs); CHECK_CONT(qs); CHECK_CONT(xq);
    const int ff = (int)qs.size(1) / 2;
    const int n = (int)qs.size(2) * 2;
    TORCH_CHECK(ff % 32 == 0, "fused grouped swiglu needs ff % 32 == 0");
    const dim3 grid(ff / 32, n_slots);
    const dim3 block(16 * WAVE);
    auto launch = [&](auto k) {
        hipLaunchKernelGGL(k, grid, block, 0, cur_stream(),
                           qs.data_ptr<uint8_t>(),
                           reinterpret_cast<const __half *>(scales.data_ptr<at::Half>()),
                           xq.data_ptr<int8_t>(), xs.data_ptr<float>(),
                           xbs.data_ptr<float>(), ff, n,
                           router.data_ptr<float>(), (int)router.size(-1),
                           (int)topk, oq.data_ptr<int8_t>(),
                           os.data_ptr<float>(), obs.data_ptr<float>());
    };
    if (gelu) launch(k_q40_gemv_grouped_swiglu<true>);
    else launch(k_q40_gemv_grouped_swiglu<false>);
}

void rope_kv_qknorm(torch::Tensor qkv, int64_t ld, int64_t q_dim0,
                    int64_t kv_dim0, torch::Tensor cache, torch::Tensor pos,
                    torch::Tensor kc, torch::Tensor vc, int64_t head_dim,
                    torch::Tensor wq, torch::Tensor wk, double eps,
                    int64_t batch) {
    CHECK_CUDA(qkv);
    const int heads = (int)((q_dim0 + 2 * kv_dim0) / head_dim);
    const dim3 grid(heads, batch);
    auto run = [&](auto kvt, auto kcp, auto vcp) {
        using KVT = decltype(kvt);
        hipLaunchKernelGGL((k_rope_kv_qknorm<KVT>), grid, dim3(WAVE), 0,
                           cur_stream(), qkv.data_ptr<float>(), (int)ld,
                           (int)q_dim0, (int)kv_dim0, cache.data_ptr<float>(),
                           pos.data_ptr<int>(), kcp, vcp, (int)head_dim,
                           wq.data_ptr<float>(), wk.data_ptr<float>(),
                           (float)eps);
    };
    if (kc.scalar_type() == at::kHalf)
        run(__half{}, reinterpret_cast<__half *>(kc.data_ptr<at::Half>()),
            reinterpret_cast<__half *>(vc.data_ptr<at::Half>()));
    else
        run(0.0f, kc.data_ptr<float>(), vc.data_ptr<float>());
}

void silu_mul(torch::Tensor a, torch::Tensor g, torch::Tensor out) {
    CHECK_CUDA(a);
    const int64_t n = a.numel();
    hipLaunchKernelGGL(k_silu_mul, dim3(ceil_div(n, 256)), dim3(256), 0, cur_stream(),
                       a.data_ptr<float>(), g.data_ptr<float>(), out.data_ptr<float>(), n);
}

// ------------------------------------------------------------ CPU Q40 GEMM
// Native CPU matmul streaming Q40 planes directly (no f32 dequant copy):
// y[b, row] = sum_blk scale * sum_k (q-8) * x — keeps CPU serving at
// quantized-weight RAM (role of the reference's AVX512/NEON CPU kernels,
// src/nn/nn-cpu-ops.cpp:231-449). Parallel over rows via at::parallel_for
// (torch threads, the CLI's --nthreads). The inner 32-wide block loop
// auto-vectorizes; activations stay f32 (more precise than the reference's
// Q80 activations, same wire format compatibility).
void q40_matmul_cpu(torch::Tensor qs, torch::Tensor scales, torch::Tensor x,
                    torch::Tensor y) {
    TORCH_CHECK(!qs.is_cuda() && !x.is_cuda(), "q40_matmul_cpu is CPU-only");
    CHECK_CONT(qs); CHECK_CONT(x);
    const int d = qs.size(0);
    const int n = qs.size(1) * 2;
    const int B = x.size(0);
    const int nb = n / QB;
    TORCH_CHECK(x.size(-1) == n, "x width mismatch");
    const uint8_t *W = qs.data_ptr<uint8_t>();
    const at::Half *S = scales.data_ptr<at::Half>();
    const float *X = x.data_ptr<float>();
    float *Y = y.data_ptr<float>();
    // Q80-quantize the activations once (the same per-32-block wire the GPU
    // path and the reference use, nn-quants.cpp:67), then int8 block dots —
    // 8-bit integer MACs are what x86 actually vectorizes for this format
    // (role of the reference's AVX2/AVX512 matmul, nn-cpu-ops.cpp:231-449).
    std::vector<int8_t> xq((size_t)B * n);
    std::vector<float> xsc((size_t)B * nb), xbs((size_t)B * nb);
    for (int b = 0; b < B; b++) {
        const float *xr = X + (int64_t)b * n;
        for (int blk = 0; blk < nb; blk++) {
            float amax = 0.0f;
            for (int j = 0; j < QB; j++)
                amax = std::max(amax, std::fabs(xr[blk * QB + j]));
            const float dd = amax / 127.0f;
            const float qinv = dd > 0.0f ? 1.0f / dd : 0.0f;
            float bsum = 0.0f;
            for (int j = 0; j < QB; j++) {
                const float qf = std::rint(xr[blk * QB + j] * qinv);
                xq[(size_t)b * n + blk * QB + j] = (int8_t)qf;
                bsum += qf;
            }
            xsc[(size_t)b * nb + blk] = dd;
            xbs[(size_t)b * nb + blk] = bsum;
        }
    }
    // OpenMP over rows: at::parallel_for silently serialized here (measured
    // 1-thread == 8-thread) and per-call std::thread spawn costs ~0.3 ms;
    // omp's persistent pool respects torch's thread count via num_threads.
    // blocktime MUST be short: libomp's default 200 ms post-region spin
    // starves torch's own intra-op pool between our regions (measured
    // 61 s/token on a 256-core box); 1 ms still covers back-to-back
    // matmuls. Capped at 64 threads — the op is memory-bound.
#if defined(_OPENMP)
    static const bool _bt0 = [] { kmp_set_blocktime(1); return true; }();
    (void)_bt0;
#endif
    const int nt = std::min(64, std::max(1, (int)at::get_num_threads()));
    auto worker = [&](int64_t r0, int64_t r1) {
        for (int64_t row = r0; row < r1; row++) {
            const uint8_t *w = W + row * (n >> 1);
            const at::Half *sc = S + row * nb;
            for (int b = 0; b < B; b++) {
                const int8_t *xr = xq.data() + (size_t)b * n;
                const float *sx = xsc.data() + (size_t)b * nb;
                const float *bs = xbs.data() + (size_t)b * nb;
                float acc = 0.0f;
#if defined(__AVX2__)
                const __m128i m0f = _mm_set1_epi8(0x0F);
                const __m256i ones = _mm256_set1_epi16(1);
                for (int blk = 0; blk < nb; blk++) {
                    const __m128i raw = _mm_loadu_si128(
                        reinterpret_cast<const __m128i *>(w + blk * 16));
                    const __m128i lo = _mm_and_si128(raw, m0f);
                    const __m128i hi = _mm_and_si128(_mm_srli_epi16(raw, 4), m0f);
                    const __m256i w8 = _mm256_set_m128i(hi, lo);  // u8 in 0..15
                    const __m256i x8 = _mm256_loadu_si256(
                        reinterpret_cast<const __m256i *>(xr + blk * QB));
                    // u8*i8 pair-sums (|pair| <= 2*15*127 < 2^15: no saturation)
                    const __m256i p16 = _mm256_maddubs_epi16(w8, x8);
                    const __m256i p32 = _mm256_madd_epi16(p16, ones);
                    __m128i h = _mm_add_epi32(_mm256_castsi256_si128(p32),
                                              _mm256_extracti128_si256(p32, 1));
                    h = _mm_add_epi32(h, _mm_shuffle_epi32(h, 0x4E));
                    h = _mm_add_epi32(h, _mm_shuffle_epi32(h, 0xB1));
                    const int idot = _mm_cvtsi128_si32(h);
                    acc += (float)sc[blk] * sx[blk]
                           * ((float)idot - 8.0f * bs[blk]);
                }
#else
                for (int blk = 0; blk < nb; blk++) {
                    const uint8_t *wb = w + blk * 16;
                    const int8_t *xb = xr + blk * QB;
                    int idot = 0;
                    for (int k = 0; k < 16; k++) {
                        idot += (int)(wb[k] & 15) * xb[k]
                              + (int)(wb[k] >> 4) * xb[k + 16];
                    }
                    acc += (float)sc[blk] * sx[blk]
                           * ((float)idot - 8.0f * bs[blk]);
                }
#endif
                Y[(int64_t)b * d + row] = acc;
            }
        }
    };
    if (nt <= 1 || d < 256) {
        worker(0, d);
    } else {
#if defined(_OPENMP)
        #pragma omp parallel num_threads(nt)
        {
            const int t = omp_get_thread_num();
            const int tn = omp_get_num_threads();
            const int64_t chunk = (d + tn - 1) / tn;
            const int64_t r0 = (int64_t)t * chunk;
            if (r0 < d) worker(r0, std::min<int64_t>(d, r0 + chunk));
        }
#else
        std::vector<std::thread> threads;
        const int64_t chunk = (d + nt - 1) / nt;
        for (int t = 0; t < nt; t++) {
            const int64_t r0 = t * chunk;
            if (r0 >= d) break;
            threads.emplace_back(worker, r0, std::min<int64_t>(d, r0 + chunk));
        }
        for (auto &th : threads) th.join();
#endif
    }
}

void sync_pack(torch::Tensor q, torch::Tensor s, torch::Tensor buf) {
    CHECK_CUDA(q);
    const int n = q.size(-1);
    const int rows = q.numel() / n;
    hipLaunchKernelGGL(k_sync_pack, dim3(ceil_div((int64_t)rows * n, 256)), dim3(256),
                       0, cur_stream(), q.data_ptr<int8_t>(), s.data_ptr<float>(),
                       buf.data_ptr<uint8_t>(), n, rows);
}

void sync_quant_pack(torch::Tensor x, torch::Tensor buf) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int rows = x.numel() / n;
    const int blocks = rows * (n / QB);
    hipLaunchKernelGGL(k_sync_quant_pack, dim3(ceil_div(blocks * 32, 256)),
                       dim3(256), 0, cur_stream(), x.data_ptr<float>(),
                       buf.data_ptr<uint8_t>(), n, blocks);
}

void scale_merge_pack(torch::Tensor y, torch::Tensor wts, torch::Tensor wire,
                      int64_t batch, int64_t topk, int64_t n) {
    CHECK_CUDA(y);
    TORCH_CHECK(n % 256 == 0, "scale_merge_pack needs dim % 256 == 0");
    hipLaunchKernelGGL(k_scale_merge_pack, dim3(n / 256, batch), dim3(256), 0,
                       cur_stream(), y.data_ptr<float>(), wts.data_ptr<float>(),
                       wire.data_ptr<uint8_t>(), (int)n, (int)topk);
}

void merge_add_q(torch::Tensor x, torch::Tensor bufs, torch::Tensor ssq,
                 torch::Tensor wnorm, torch::Tensor oq, torch::Tensor os,
                 torch::Tensor obs) {
    // bufs [world, rows, row_bytes] (all-gather output); x [rows, n]
    CHECK_CUDA(x); CHECK_CONT(bufs);
    const int world = bufs.size(0);
    const int rows = bufs.size(1);
    const int n = x.size(-1);
    TORCH_CHECK(n % 256 == 0, "merge_add_q needs dim % 256 == 0");
    hipLaunchKernelGGL(k_merge_add_q, dim3(n / 256, rows), dim3(256), 0,
                       cur_stream(), x.data_ptr<float>(),
                       bufs.data_ptr<uint8_t>(), ssq.data_ptr<float>(),
                       wnorm.data_ptr<float>(), oq.data_ptr<int8_t>(),
                       os.data_ptr<float>(), obs.data_ptr<float>(),
                       world, n, rows);
}

void merge_add(torch::Tensor x, torch::Tensor bufs,
               c10::optional<torch::Tensor> ssq = c10::nullopt) {
    CHECK_CUDA(x);
    const int n = x.size(-1);
    const int rows = x.numel() / n;
    const int world = bufs.size(0);
    hipLaunchKernelGGL(k_merge_add, dim3(ceil_div(n, 1024), rows), dim3(256),
                       0, cur_stream(), x.data_ptr<float>(), bufs.data_ptr<uint8_t>(),
                       ssq.has_value() ? ssq->data_ptr<float>() : nullptr,
                       world, n, rows);
}

void add_(torch::Tensor x, torch::Tensor y) {
    CHECK_CUDA(x);
    hipLaunchKernelGGL(k_add, dim3(ceil_div(x.numel(), 256)), dim3(256), 0,
                       cur_stream(), x.data_ptr<float>(), y.data_ptr<float>(), x.numel());
}

void pos_inc(torch::Tensor pos, int64_t delta) {
    hipLaunchKernelGGL(k_inc, dim3(1), dim3(64), 0, cur_stream(),
                       pos.data_ptr<int>(), (int)delta);
}

// ===================================================== native BPE encoder
// Host-side tokenizer encode (the reference's tokenizer is native C++,
// src/tokenizer.cpp:311-390): exact-match byte accumulation with special
// token scan, then greedy highest-score pair merging. The Python Tokenizer
// delegates here when the extension is loaded.
#include <algorithm>
#include <string>
#include <unordered_map>
#include <vector>

struct BpeEncoder {
    std::vector<std::string> vocab;
    std::vector<float> scores;
    std::unordered_map<std::string, int> regular;  // first-id wins
    std::vector<int> special_ids;                  // ids >= regular_size
    int regular_size;

    BpeEncoder(const std::vector<std::string> &vocab_,
               const std::vector<double> &scores_, int64_t regular_size_)
        : vocab(vocab_), regular_size((int)regular_size_) {
        scores.reserve(scores_.size());
        for (double s : scores_) scores.push_back((float)s);
        for (int i = 0; i < regular_size && i < (int)vocab.size(); i++)
            regular.emplace(vocab[i], i);  // emplace keeps the first id
        for (int i = regular_size; i < (int)vocab.size(); i++)
            special_ids.push_back(i);
    }

    std::vector<int> encode(const std::string &data, bool add_special) const {
        std::vector<int> tokens;
        std::string buf;
        size_t i = 0;
        while (i < data.size()) {
            if (add_special) {
                int sp = -1;
                for (int sid : special_ids) {
                    const std::string &p = vocab[sid];
                    if (!p.empty() && data.compare(i, p.size(), p) == 0) {
                        sp = sid;
                        break;
                    }
                }
                if (sp >= 0) {
                    if (!buf.empty())
                        throw std::runtime_error("unencodable byte run before special");
                    tokens.push_back(sp);
                    i += vocab[sp].size();
                    continue;
                }
            }
            buf.push_back(data[i++]);
            auto it = regular.find(buf);
            if (it != regular.end()) {
                tokens.push_back(it->second);
                buf.clear();
            }
        }
        if (!buf.empty())
            throw std::runtime_error("cannot encode byte run");

        // greedy merge: globally best-score adjacent pair each round with
        // leftmost tie-break — identical semantics to the reference's O(n^2)
        // rescan (tokenizer.cpp:352-379) via heap + linked list + lazy
        // invalidation (O(n log n)).
        const int n = (int)tokens.size();
        if (n == 0) return tokens;
        std::vector<int> tok(tokens), prev(n), next(n), ver(n, 0);
        for (int j = 0; j < n; j++) { prev[j] = j - 1; next[j] = j + 1 < n ? j + 1 : -1; }
        struct Cand { float score; int pos, id, vl, vr, right; };
        auto worse = [](const Cand &a, const Cand &b) {
            if (a.score != b.score) return a.score < b.score;  // max-heap
            return a.pos > b.pos;                              // leftmost wins
        };
        std::vector<Cand> heap;
        auto push_pair = [&](int l) {
            const int r = next[l];
            if (l < 0 || r < 0) return;
            auto it = regular.find(vocab[tok[l]] + vocab[tok[r]]);
            if (it == regular.end()) return;
            heap.push_back({scores[it->second], l, it->second, ver[l], ver[r], r});
            std::push_heap(heap.begin(), heap.end(), worse);
        };
        for (int j = 0; j + 1 < n; j++) push_pair(j);
        while (!heap.empty()) {
            std::pop_heap(heap.begin(), heap.end(), worse);
            const Cand c = heap.back();
            heap.pop_back();
            const int l = c.pos, r = c.right;
            if (ver[l] != c.vl || ver[r] != c.vr || next[l] != r) continue;
            tok[l] = c.id;
            ver[l]++;
            ver[r]++;
            next[l] = next[r];
            if (next[r] >= 0) prev[next[r]] = l;
            push_pair(prev[l] >= 0 ? prev[l] : -1);
            push_pair(l);
        }
        std::vector<int> out;
        for (int j = 0; j >= 0; j = next[j]) out.push_back(tok[j]);
        return out;
    }
};

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    py::class_<BpeEncoder>(m, "BpeEncoder")
        .def(py::init<const std::vector<std::string> &,
                      const std::vector<double> &, int64_t>())
        .def("encode", [](const BpeEncoder &e, const py::bytes &data,
                          bool add_special) {
            return e.encode(std::string(data), add_special);
        });
    m.def("q80_quantize", &q80_quantize);
    m.def("rmsnorm", &rmsnorm);
    m.def("rmsnorm_q80", &rmsnorm_q80);
    m.def("rmsnorm_rows", &rmsnorm_rows);
    m.def("rmsnorm_rows_s", &rmsnorm_rows_s);
    m.def("add_rmsnorm", &add_rmsnorm);
    m.def("add_rmsnorm_q80", &add_rmsnorm_q80);
    m.def("q40_gemv", &q40_gemv, py::arg("qs"), py::arg("scales"), py::arg("xq"),
          py::arg("xs"), py::arg("xbs"), py::arg("y"), py::arg("batch"),
          py::arg("amax_slot") = py::none(), py::arg("ssq_in") = py::none(),
          py::arg("eps") = 0.0);
    m.def("q40_gemv_resid", &q40_gemv_resid);
    m.def("q40_gemv_resid_q", &q40_gemv_resid_q);
    m.def("q40_gemv_pack", &q40_gemv_pack);
    m.def("q40_gemv_ksplit", &q40_gemv_ksplit);
    m.def("add_ssq_q", &add_ssq_q);
    m.def("merge_add_q", &merge_add_q);
    m.def("scale_merge_pack", &scale_merge_pack);
    m.def("q40_matmul_cpu", &q40_matmul_cpu);
    m.def("q40_gemm", &q40_gemm, py::arg("qs"), py::arg("scales"),
          py::arg("xq"), py::arg("xs"), py::arg("y"), py::arg("batch"),
          py::arg("part") = py::none(), py::arg("variant") = -1);
    m.def("q40_gemv_rope", &q40_gemv_rope, py::arg("qs"), py::arg("scales"),
          py::arg("xq"), py::arg("xs"), py::arg("xbs"), py::arg("y"),
          py::arg("batch"), py::arg("cache"), py::arg("pos"), py::arg("kc"),
          py::arg("vc"), py::arg("q_dim0"), py::arg("kv_dim0"),
          py::arg("head_dim"), py::arg("ssq_in") = py::none(),
          py::arg("eps") = 0.0);
    m.def("q40_gemv_swiglu", &q40_gemv_swiglu, py::arg("qs"), py::arg("scales"),
          py::arg("xq"), py::arg("xs"), py::arg("xbs"), py::arg("oq"),
          py::arg("os"), py::arg("obs"), py::arg("gelu") = false);
    m.def("q40_gemv_nq", &q40_gemv_nq, py::arg("qs"), py::arg("scales"),
          py::arg("x"), py::arg("wnorm"), py::arg("ssq"), py::arg("eps"),
          py::arg("y"), py::arg("batch"), py::arg("amax_slot") = py::none());
    m.def("q40_gemv_nq_rope", &q40_gemv_nq_rope);
    m.def("norm_quant", &norm_quant, py::arg("x"), py::arg("w"), py::arg("ssq"),
          py::arg("q"), py::arg("s"), py::arg("bs"), py::arg("batch"),
          py::arg("eps"), py::arg("yout") = py::none(),
          py::arg("deferred") = false);
    m.def("moe_gate", &moe_gate);
    m.def("scale_merge_add", &scale_merge_add, py::arg("x"), py::arg("y"),
          py::arg("wts"), py::arg("ssq"), py::arg("batch"), py::arg("topk"),
          py::arg("gate") = false);
    m.def("scale_merge", &scale_merge, py::arg("partial"), py::arg("y"),
          py::arg("wts"), py::arg("batch"), py::arg("topk"),
          py::arg("gate") = false);
    m.def("logits_concat", &logits_concat);
    m.def("argmax_token", &argmax_token);
    m.def("router_gemv", &router_gemv);
    m.def("router_gemv_norm", &router_gemv_norm);
    m.def("scale_merge_add_q", &scale_merge_add_q);
    m.def("norm_f32", &norm_f32);
    m.def("add_ssq", &add_ssq);
    m.def("q40_gemv_grouped", &q40_gemv_grouped, py::arg("qs"),
          py::arg("scales"), py::arg("xq"), py::arg("xs"), py::arg("xbs"),
          py::arg("expert_idx"), py::arg("y"), py::arg("k_slots"),
          py::arg("variant") = -1, py::arg("router") = py::none(),
          py::arg("topk") = 0, py::arg("n_slots") = 0,
          py::arg("ssq_in") = py::none(), py::arg("eps") = 0.0);
    m.def("q40_gemv_grouped_swiglu", &q40_gemv_grouped_swiglu, py::arg("qs"),
          py::arg("scales"), py::arg("xq"), py::arg("xs"), py::arg("xbs"),
          py::arg("oq"), py::arg("os"), py::arg("obs"), py::arg("n_slots"),
          py::arg("router"), py::arg("topk"), py::arg("gelu") = false);
    m.def("rope_kv_qknorm", &rope_kv_qknorm);
    m.def("rope", &rope);
    m.def("rope_kv", &rope_kv);
    m.def("kv_append", &kv_append);
    m.def("attn", &attn, py::arg("q"), py::arg("q_ld"), py::arg("kc"), py::arg("vc"),
          py::arg("y"), py::arg("pos"), py::arg("batch"), py::arg("n_heads0"),
          py::arg("kv_mul"), py::arg("head_dim"), py::arg("splits"),
          py::arg("ml_scratch"), py::arg("o_scratch"), py::arg("counter"),
          py::arg("zq") = py::none(), py::arg("zs") = py::none(),
          py::arg("zbs") = py::none());
    m.def("embed_gather", &embed_gather, py::arg("table"), py::arg("tokens"), py::arg("x"), py::arg("batch"), py::arg("ssq") = py::none());
    m.def("swiglu_q80", &swiglu_q80, py::arg("a"), py::arg("g"),
          py::arg("lda"), py::arg("n"), py::arg("rows"), py::arg("q"),
          py::arg("s"), py::arg("bs"), py::arg("gelu") = false);
    m.def("silu_mul", &silu_mul);
    m.def("sync_pack", &sync_pack);
    m.def("sync_quant_pack", &sync_quant_pack);
    m.def("merge_add", &merge_add, py::arg("x"), py::arg("bufs"), py::arg("ssq") = py::none());
    m.def("add_", &add_);
    m.def("pos_inc", &pos_inc);
    m.def("token_from_argmax", &token_from_argmax);
    m.def("q40_gemv_argmax_blocks", &q40_gemv_argmax_blocks);
}
