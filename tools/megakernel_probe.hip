// Round-2 decode "megakernel" feasibility probe (NOT part of the runtime).
//
// Decode spends ~1.6 us of in-graph launch floor per kernel x ~8 kernels
// per layer (profiles/r01_decode_llama31_8b.md). A fused per-layer
// megakernel would replace 7 kernel boundaries with 7 grid-wide barriers —
// worth it only if a grid barrier costs well under the launch floor.
// This probe measures:
//   1. per-launch cost of a 256-kernel hipGraph chain (the known floor)
//   2. per-sync cost of cooperative-groups grid.sync()
//   3. per-sync cost of a hand-rolled generation barrier (works inside
//      hipGraphs, unlike cooperative launch on some stacks)
// Build: hipcc --offload-arch=gfx950 -O3 tools/megakernel_probe.hip -o /tmp/mk
// Run (GPU box): /tmp/mk
#include <hip/hip_runtime.h>
#include <hip/hip_cooperative_groups.h>
#include <cstdio>
#include <cstdlib>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    exit(1); } } while (0)

__global__ void k_empty(float *p) {
    if (p && threadIdx.x == 0 && blockIdx.x == 0) p[0] += 1.0f;
}

__global__ void k_gridsync(float *p, int iters) {
    namespace cg = cooperative_groups;
    cg::grid_group g = cg::this_grid();
    for (int i = 0; i < iters; i++) {
        g.sync();
        if (p && threadIdx.x == 0 && blockIdx.x == 0) p[0] += 1.0f;
    }
}

// generation-counter barrier: every workgroup must be resident (size the
// grid from the occupancy API or this deadlocks)
__device__ __forceinline__ void soft_barrier(unsigned *count, unsigned *gen,
                                             int nwg) {
    __syncthreads();
    if (threadIdx.x == 0) {
        __threadfence();
        const unsigned g = atomicAdd(gen, 0u);
        if (atomicAdd(count, 1u) == (unsigned)(nwg - 1)) {
            atomicExch(count, 0u);
            __threadfence();
            atomicAdd(gen, 1u);
        } else {
            while (atomicAdd(gen, 0u) == g)
                __builtin_amdgcn_s_sleep(8);
        }
    }
    __syncthreads();
}

__global__ void k_softsync(float *p, unsigned *count, unsigned *gen, int iters) {
    const int nwg = gridDim.x;
    for (int i = 0; i < iters; i++) {
        soft_barrier(count, gen, nwg);
        if (p && threadIdx.x == 0 && blockIdx.x == 0) p[0] += 1.0f;
    }
}

int main() {
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, 0));
    printf("device: %s, %d CUs\n", prop.gcnArchName, prop.multiProcessorCount);
    float *dp;
    HIP_CHECK(hipMalloc(&dp, 4));
    HIP_CHECK(hipMemset(dp, 0, 4));
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));

    // 1) graph of 256 empty kernels
    {
        hipStream_t s;
        HIP_CHECK(hipStreamCreate(&s));
        hipGraph_t graph;
        hipGraphExec_t exec;
        HIP_CHECK(hipStreamBeginCapture(s, hipStreamCaptureModeGlobal));
        for (int i = 0; i < 256; i++)
            hipLaunchKernelGGL(k_empty, dim3(1024), dim3(256), 0, s, dp);
        HIP_CHECK(hipStreamEndCapture(s, &graph));
        HIP_CHECK(hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0));
        HIP_CHECK(hipGraphLaunch(exec, s));
        HIP_CHECK(hipStreamSynchronize(s));
        HIP_CHECK(hipEventRecord(e0, s));
        for (int r = 0; r < 20; r++) HIP_CHECK(hipGraphLaunch(exec, s));
        HIP_CHECK(hipEventRecord(e1, s));
        HIP_CHECK(hipStreamSynchronize(s));
        float ms;
        HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
        printf("graph kernel-launch floor: %.2f us/kernel\n",
               ms * 1000.0 / (20.0 * 256));
    }

    // 2) cooperative grid.sync
    {
        int supported = 0;
        HIP_CHECK(hipDeviceGetAttribute(&supported,
                                        hipDeviceAttributeCooperativeLaunch, 0));
        if (!supported) {
            printf("cooperative launch: NOT supported\n");
        } else {
            int maxBlocks = 0;
            HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
                &maxBlocks, k_gridsync, 256, 0));
            const int grid = maxBlocks * prop.multiProcessorCount;
            const int iters = 2000;
            int it = iters;
            void *args[] = {&dp, &it};
            HIP_CHECK(hipLaunchCooperativeKernel((const void *)k_gridsync,
                                                 dim3(grid), dim3(256), args, 0, 0));
            HIP_CHECK(hipDeviceSynchronize());
            HIP_CHECK(hipEventRecord(e0));
            HIP_CHECK(hipLaunchCooperativeKernel((const void *)k_gridsync,
                                                 dim3(grid), dim3(256), args, 0, 0));
            HIP_CHECK(hipEventRecord(e1));
            HIP_CHECK(hipEventSynchronize(e1));
            float ms;
            HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
            printf("cooperative grid.sync (%d wg): %.2f us/sync\n", grid,
                   ms * 1000.0 / iters);
        }
    }

    // 3) hand-rolled barrier (normal launch, resident grid), eager + graph
    {
        unsigned *ctrs;
        HIP_CHECK(hipMalloc(&ctrs, 8));
        HIP_CHECK(hipMemset(ctrs, 0, 8));
        int maxBlocks = 0;
        HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &maxBlocks, k_softsync, 256, 0));
        const int grid = maxBlocks * prop.multiProcessorCount;
        const int iters = 2000;
        hipLaunchKernelGGL(k_softsync, dim3(grid), dim3(256), 0, 0,
                           dp, ctrs, ctrs + 1, iters);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_softsync, dim3(grid), dim3(256), 0, 0,
                           dp, ctrs, ctrs + 1, iters);
        HIP_CHECK(hipEventRecord(e1));
        HIP_CHECK(hipEventSynchronize(e1));
        float ms;
        HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
        printf("soft barrier (%d wg): %.2f us/sync\n", grid, ms * 1000.0 / iters);

        // inside a graph (cooperative launches often can't be captured;
        // the soft barrier can)
        hipStream_t s;
        HIP_CHECK(hipStreamCreate(&s));
        hipGraph_t graph;
        hipGraphExec_t exec;
        HIP_CHECK(hipStreamBeginCapture(s, hipStreamCaptureModeGlobal));
        hipLaunchKernelGGL(k_softsync, dim3(grid), dim3(256), 0, s,
                           dp, ctrs, ctrs + 1, iters);
        HIP_CHECK(hipStreamEndCapture(s, &graph));
        HIP_CHECK(hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0));
        HIP_CHECK(hipGraphLaunch(exec, s));
        HIP_CHECK(hipStreamSynchronize(s));
        HIP_CHECK(hipEventRecord(e0, s));
        HIP_CHECK(hipGraphLaunch(exec, s));
        HIP_CHECK(hipEventRecord(e1, s));
        HIP_CHECK(hipStreamSynchronize(s));
        HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
        printf("soft barrier in hipGraph: %.2f us/sync\n", ms * 1000.0 / iters);
    }
    return 0;
}
