"""Per-rank HBM sizing audit (BASELINE config 5: Llama-3.1-405B Q40 on one
8x288GB MI355X node).

Accounts every device allocation the HIP runtime makes (weights in GEMV
plane layout, f32 KV cache, activation buffers, sync buffers) and checks
the shard fits; run: python tools/sizing_audit.py [seq_len]."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from dllama_amd import model_file as mf
from dllama_amd.models.config import ModelConfig

HBM = 288e9
NB = 32


def q40_plane_bytes(d, n):
    return d * n // 2 + d * (n // 32) * 2  # nibbles + f16 scales


def audit(name: str, world: int, seq_len: int):
    h = mf.preset_header(name, seq_len=seq_len)
    c = ModelConfig.from_header(h, world=world)
    per_layer_w = (q40_plane_bytes(c.q_dim0 + 2 * c.kv_dim0, c.dim)   # qkv
                   + q40_plane_bytes(c.dim, c.q_dim0))                # wo
    if c.is_moe:
        per_layer_w += c.n_experts * (q40_plane_bytes(2 * c.ff_dim0, c.dim)
                                      + q40_plane_bytes(c.dim, c.ff_dim0))
        per_layer_w += c.n_experts * c.dim * 4  # gate f32
    else:
        per_layer_w += (q40_plane_bytes(2 * c.ff_dim0, c.dim)
                        + q40_plane_bytes(c.dim, c.ff_dim0))
    per_layer_w += 2 * c.dim * 4  # norms
    weights = (c.n_layers * per_layer_w
               + c.vocab_size * c.dim * 4           # f32 embedding table
               + q40_plane_bytes(c.vocab0, c.dim)   # logits shard
               + c.dim * 4)
    kv = c.n_layers * seq_len * c.kv_dim0 * 2 * 2  # f16 KV (round-2 default)
    act = NB * (3 * c.dim + c.q_dim0 + 2 * c.kv_dim0 + 2 * c.q_dim0
                + 4 * c.ff_dim0 + c.vocab0) * 4
    act += NB * (c.dim + c.q_dim0 + c.ff_dim0) * 2  # int8+scale quant bufs
    if c.is_moe:
        act += NB * c.n_active_experts * (3 * c.ff_dim0 + c.dim) * 4
    sync = world * NB * (c.dim + c.dim // 16) * 2 if world > 1 else 0
    total = weights + kv + act + sync
    print(f"{name:16s} TP={world}  seq={seq_len}")
    print(f"  weights/rank: {weights/1e9:8.2f} GB")
    print(f"  kv cache    : {kv/1e9:8.2f} GB (f16, {seq_len} x {c.kv_dim0} x {c.n_layers}L x 2)")
    print(f"  activations : {act/1e9:8.2f} GB   sync: {sync/1e6:.0f} MB")
    fits = "✅ fits" if total < HBM * 0.97 else "❌ DOES NOT FIT"
    print(f"  total/rank  : {total/1e9:8.2f} GB of 288 GB  -> {fits}\n")
    return total


if __name__ == "__main__":
    seq = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    audit("llama-3.1-8b", 1, seq)
    audit("llama-3.3-70b", 8, seq)
    audit("qwen3-30b-a3b", 4, seq)
    audit("llama-3.1-405b", 8, seq)
    audit("llama-3.1-405b", 8, 32768)
    audit("llama-3.1-405b", 1, 4096)
