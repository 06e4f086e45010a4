"""TP-machinery floor on one GPU: world=1 nccl process group, force_sync
decode (Q80 pack -> real RCCL all-gather -> merge-add, 64 collectives +
logits gather per step) captured in the decode graph, vs the plain path.
The delta bounds the per-step cost the 8-GPU run adds BEFORE any xGMI wire
time (RCCL kernel launches, pack/merge kernels, gather+concat)."""
import os, sys, time
sys.path.insert(0, "/root/repo")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29591")
os.environ["WORLD_SIZE"] = "1"
os.environ["RANK"] = "0"
os.environ["LOCAL_RANK"] = "0"
import torch
import torch.distributed as dist
dist.init_process_group("nccl", rank=0, world_size=1)
from dllama_amd.model_file import preset_header
from dllama_amd.models.config import ModelConfig
from dllama_amd.models.hip_model import HipTransformer
from dllama_amd.parallel.comm import DistComm

h = preset_header("llama-3.1-8b", seq_len=4096)
cfg = ModelConfig.from_header(h)

def run(force_sync):
    m = HipTransformer.synthetic(cfg, comm=DistComm() if force_sync else None,
                                 force_sync=force_sync)
    m.greedy_feedback = True
    m.forward(torch.randint(0, 128256, (32,)), torch.arange(32))
    m.capture_decode_graph()
    m.pos.fill_(32)
    m.tokens[0] = 7
    for _ in range(30):
        m._graph.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(200):
        m._graph.replay()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 200
    del m
    torch.cuda.empty_cache()
    return dt

plain = run(False)
tp = run(True)
print(f"plain decode:        {plain*1e3:.3f} ms/step ({1/plain:.1f} tok/s)")
print(f"TP path (rccl w=1):  {tp*1e3:.3f} ms/step ({1/tp:.1f} tok/s)")
print(f"TP machinery floor:  {(tp-plain)*1e6:.0f} us/step over 65 collectives "
      f"({(tp-plain)*1e6/65:.2f} us each incl pack+merge)")
dist.destroy_process_group()
