#!/usr/bin/env python3
"""Cross-validation against the reference C++ implementation.

Builds b4rtaz/distributed-llama's `dllama` from /root/reference (pure C++,
CPU), writes a tiny synthetic model + ASCII-only tokenizer with THIS
framework's .m/.t writers, then greedy-decodes the same prompt with both
runtimes and compares the generated text exactly.

This proves end-to-end: file-format byte compatibility, weight-walk order,
Q40/Q80 quantization parity, and model-math parity (same argmax trajectory).
"""

import os
import shutil
import subprocess
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from dllama_amd import model_file as mf
from dllama_amd import tokenizer as tk

REF_SRC = "/root/reference"


def build_reference(workdir: str) -> str:
    ref = os.path.join(workdir, "refbuild")
    if not os.path.exists(ref):
        shutil.copytree(REF_SRC, ref)
    binary = os.path.join(ref, "dllama")
    if not os.path.exists(binary):
        subprocess.run(["make", "dllama"], cwd=ref, check=True,
                       capture_output=True)
    return binary


def make_ascii_assets(workdir: str, vocab_size: int = 128, arch: str = "llama"):
    """Tokenizer whose every regular token is printable ASCII -> decode is
    lossless text on both sides."""
    printable = [chr(c).encode() for c in range(0x20, 0x7F)]  # 95 tokens
    filler = [b"~"] * (vocab_size - 5 - len(printable))       # dup ids unused
    vocab = printable + filler
    scores = [0.0] * len(vocab)
    bos_id = len(vocab)
    vocab += [b"<|begin|>", b"<|end|>", b"<|h|>", b"<|e|>", b"<|eot|>"]
    scores += [0.0] * 5
    tok_path = os.path.join(workdir, "parity.t")
    tk.write_tokenizer(tok_path, vocab, scores, bos_id, True, [bos_id + 1],
                       "{{<|start_header_id|>}}")

    if arch == "qwen3_moe":
        h = mf.LlmHeader(arch_type=mf.ARCH_QWEN3_MOE, dim=64, hidden_dim=128,
                         n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                         n_experts=4, n_active_experts=2, moe_hidden_dim=64,
                         vocab_size=vocab_size, seq_len=256, rope_theta=10000,
                         norm_epsilon=1e-6)
    elif arch == "qwen3":
        h = mf.LlmHeader(arch_type=mf.ARCH_QWEN3, dim=64, hidden_dim=128,
                         n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                         vocab_size=vocab_size, seq_len=256, rope_theta=10000,
                         norm_epsilon=1e-6)
    else:
        h = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=64, hidden_dim=128,
                         n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                         vocab_size=vocab_size, seq_len=256, rope_theta=10000,
                         rope_type=mf.ROPE_LLAMA)
    h.finalize()
    model_path = os.path.join(workdir, f"parity_{arch}.m")
    write_structured_model(model_path, h)
    return model_path, tok_path


def write_structured_model(path: str, h: mf.LlmHeader, seed: int = 11) -> None:
    """Random model EXCEPT wcls[i] = embedding[i-1]: logits argmax is
    'input token + 1' with decisive margins, so greedy trajectories are
    robust to rounding-mode differences between runtimes (a fully random
    model has near-flat logits and argmax ties flip on 1-ulp noise)."""
    import numpy as np
    rng = np.random.default_rng(seed)
    emb = rng.standard_normal((h.vocab_size, h.dim)).astype(np.float32) * 0.03
    with open(path, "wb") as f:
        mf.write_header(f, h)
    hdr = mf.read_header(path)
    with open(path, "ab") as f:
        for e in mf.tensor_walk(hdr):
            if e.name == "embedding":
                x = emb
            elif e.name == "final_matmul_logits":
                x = np.roll(emb, 1, axis=0)  # wcls[i] = emb[i-1]
            elif e.name in ("block_norm_0", "block_norm_1", "final_norm"):
                x = np.ones(int(np.prod(e.shape)), dtype=np.float32)
            else:
                x = rng.standard_normal(int(np.prod(e.shape))).astype(np.float32) * 0.03
            mf.write_tensor(f, x, e.float_type)


def run_reference(binary, model, tok, prompt, steps) -> str:
    out = subprocess.run(
        [binary, "inference", "--model", model, "--tokenizer", tok,
         "--prompt", prompt, "--steps", str(steps), "--temperature", "0",
         "--nthreads", "2", "--buffer-float-type", "q80",
         "--max-seq-len", "256"],
        capture_output=True, text=True, timeout=300, check=True)
    pieces = []
    for line in out.stdout.splitlines():
        if line.startswith("🔶"):
            pieces.append(line.rsplit("| ", 1)[1] if "| " in line else "")
    return "".join(pieces)


def run_ours(model, tok_path, prompt, steps) -> str:
    """Greedy-decode with the REFERENCE's prompt handoff emulated.

    The reference inference loop has an off-by-one (dllama.cpp:56 reads
    `inputTokens[pos + 1]` after `pos += batchSize`), so its decode starts
    from a zero token at position n-1 instead of the last prompt token.
    To compare trajectories we reproduce exactly that: prefill
    tokens[:-1] + [0], then sample steps-n+1 tokens.
    (dllama_amd's own engine uses the correct handoff.)"""
    from dllama_amd.engine import InferenceEngine
    from dllama_amd.models.config import ModelConfig
    from dllama_amd.models.cpu_model import CpuTransformer
    from dllama_amd.tokenizer import Sampler, Tokenizer
    m = mf.ModelFile(model, max_seq_len=256)
    tok = Tokenizer(tok_path)
    eng = InferenceEngine(CpuTransformer(m, ModelConfig.from_header(m.header)),
                          tok, Sampler(m.header.vocab_size, 0.0, 0.9, 1))
    tokens = tok.encode(prompt)
    emulated = tokens[:-1] + [0]
    out, _ = eng.generate(emulated, steps - len(tokens) + 1)
    tok.reset_decoder()
    return "".join(p for p in (tok.decode(t) for t in out) if p)


def main():
    workdir = sys.argv[1] if len(sys.argv) > 1 else tempfile.mkdtemp(prefix="parity")
    os.makedirs(workdir, exist_ok=True)
    binary = build_reference(workdir)
    rc = 0
    for arch in ("llama", "qwen3", "qwen3_moe"):
        model, tok = make_ascii_assets(workdir, arch=arch)
        prompt = "hello world, this is"
        steps = 48
        ref_text = run_reference(binary, model, tok, prompt, steps)
        our_text = run_ours(model, tok, prompt, steps)
        print(f"[{arch}] reference: {ref_text!r}")
        print(f"[{arch}] dllama_amd:{our_text!r}")
        if ref_text == our_text:
            print(f"✅ [{arch}] PARITY: {steps} greedy tokens identical "
                  "(format + quantization + model math)")
        else:
            n = sum(1 for a, b in zip(ref_text, our_text) if a == b)
            print(f"❌ [{arch}] MISMATCH after {n} matching chars")
            rc = 1
    return rc


if __name__ == "__main__":
    sys.exit(main())
