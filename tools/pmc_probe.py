"""Small workload for rocprofv3 --pmc runs: a few eager decode steps and
one 32-token prefill chunk of the synthetic Llama-3.1-8B."""
import sys, torch
sys.path.insert(0, "/root/repo")
from dllama_amd.model_file import preset_header
from dllama_amd.models.config import ModelConfig
from dllama_amd.models.hip_model import HipTransformer

h = preset_header("llama-3.1-8b", seq_len=4096)
cfg = ModelConfig.from_header(h)
m = HipTransformer.synthetic(cfg)
m._pf_failed = True  # eager everywhere so PMC sees plain dispatches
m.forward(torch.randint(0, 128256, (32,)), torch.arange(32))   # prefill chunk
for step in range(4):                                           # decode steps
    m.forward(torch.tensor([7 + step]), torch.tensor([32 + step]))
torch.cuda.synchronize()
print("pmc workload done")
