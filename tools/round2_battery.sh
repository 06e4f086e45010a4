#!/bin/sh
# Round-2 opening GPU battery (see docs/ROUND2.md). Run on a GPU box:
#   /usr/local/graft/bin/gpurun --timeout 900 -- 'sh tools/round2_battery.sh > gpurun_out/battery.log 2>&1'
set -x
cd "$(dirname "$0")/.." || exit 1
mkdir -p gpurun_out

hipcc --offload-arch=gfx950 -O3 tools/gemm_v2_probe.hip -o /tmp/gemm_v2
hipcc --offload-arch=gfx950 -O3 tools/megakernel_probe.hip -o /tmp/mk
hipcc --offload-arch=gfx950 -O3 tools/attn_kv16_probe.hip -o /tmp/attn16
hipcc --offload-arch=gfx950 -O3 tools/moe_gemv_probe.hip -o /tmp/moe

timeout 120 /tmp/gemm_v2 14336 4096 32      # w13-shaped GEMM A/B
timeout 120 /tmp/gemm_v2 4096 4096 32       # wo-shaped
timeout 120 /tmp/gemm_v2 6144 4096 32       # qkv-shaped
timeout 120 /tmp/mk                         # launch floor vs grid barriers
timeout 120 /tmp/attn16                     # f32 vs f16 KV attention
timeout 120 /tmp/moe                        # grouped GEMV LPP sweep

DLLAMA_EXPERIMENTAL=1 timeout 600 python -m pytest tests/ -m gpu -q
timeout 120 python tools/kernel_bench.py    # incl. in-model v1-vs-v2 A/B
timeout 150 python bench.py --steps 100 --warmup 16
DLLAMA_GEMM_V2=1 timeout 150 python bench.py --steps 64 --warmup 8 --prefill 512
DLLAMA_MOE_V2=1 timeout 200 python bench.py --model qwen3-30b-a3b --steps 100 --warmup 16
