"""Prefill throughput probe: 512-token prompt through the int8-MFMA GEMM path."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from dllama_amd.model_file import preset_header
from dllama_amd.models.config import ModelConfig
from dllama_amd.models.hip_model import HipTransformer

h = preset_header("llama-3.1-8b", seq_len=4096)
cfg = ModelConfig.from_header(h)
m = HipTransformer.synthetic(cfg)
prompt = torch.randint(0, 128256, (512,))
for rep in range(3):
    m.pos.fill_(0)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for i in range(0, 512, 32):
        m.skip_logits = i < 480
        m.forward(prompt[i:i+32], torch.arange(i, i+32))
        m.skip_logits = False
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"prefill 512 tokens: {dt*1000:.1f} ms = {512/dt:.0f} tok/s")
