#!/bin/sh
# Run on N GPUs of one node (role of reference examples/n-workers.sh, which
# launched N TCP workers in screen sessions; here workers are torchrun ranks
# over RCCL/xGMI).
N=${1:-8}
[ $# -gt 0 ] && shift
torchrun --nproc-per-node "$N" --master-addr 127.0.0.1 \
    -m dllama_amd.apps.main inference "$@"
