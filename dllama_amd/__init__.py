"""dllama_amd — an MI355X-native tensor-parallel LLM inference framework.

A from-scratch re-design of the capabilities of b4rtaz/distributed-llama
(reference layer map in SURVEY.md) for a single 8xMI355X node:

- Q40 block-quantized weights / Q80 activation quantization
  (format parity with reference src/nn/nn-quants.hpp:53-72)
- hand-written CDNA4 (gfx950) HIP kernels for every hot op
- tensor parallelism across 2^n GPUs with RCCL collectives over xGMI
  (replacing the reference's TCP socket mesh, src/nn/nn-network.cpp)
- `.m` model / `.t` tokenizer file formats (reference src/llm.cpp:36-116,
  src/tokenizer.cpp:42-164)
- CLI (`dllama inference|chat|worker`) and an OpenAI-compatible API server.
"""

__version__ = "0.1.0"
