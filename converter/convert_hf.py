#!/usr/bin/env python3
"""HF safetensors checkpoint -> .m converter.

Behavior parity with the reference converter (converter/convert-hf.py):
  - same canonical tensor order as the runtime weight walk
    (dllama_amd/model_file.py tensor_walk / reference llm.cpp:614-661)
  - Llama Q/K head permutation from HF's half-rotated layout to the
    interleaved-pair rope layout (reference convert-hf.py:13-16)
  - same header keys and arch/act/rope-type mappings.

Usage: python converter/convert_hf.py <hf_folder> <q40|q80|f32> <name>
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import numpy as np

from dllama_amd import model_file as mf
from dllama_amd.quants import F32, Q40, Q80

FLOAT_TYPES = {"f32": F32, "q40": Q40, "q80": Q80}
ARCH_TYPES = {"llama": mf.ARCH_LLAMA, "mistral": mf.ARCH_LLAMA,
              "qwen3": mf.ARCH_QWEN3, "qwen3_moe": mf.ARCH_QWEN3_MOE}
ACTS = {"gelu": mf.HIDDEN_ACT_GELU, "silu": mf.HIDDEN_ACT_SILU}


def permute_qk(t: np.ndarray, n_heads: int) -> np.ndarray:
    """HF half-rotated -> interleaved-pair rope layout
    (reference convert-hf.py:13-16)."""
    d = t.shape[0]
    return (t.reshape(n_heads, 2, d // n_heads // 2, *t.shape[1:])
            .swapaxes(1, 2).reshape(t.shape))


def header_from_config(cfg: dict, weight_type: int) -> mf.LlmHeader:
    h = mf.LlmHeader(
        arch_type=ARCH_TYPES[cfg["model_type"]],
        dim=cfg["hidden_size"],
        hidden_dim=cfg["intermediate_size"],
        n_layers=cfg["num_hidden_layers"],
        n_heads=cfg["num_attention_heads"],
        n_kv_heads=cfg["num_key_value_heads"],
        vocab_size=cfg["vocab_size"],
        seq_len=cfg["max_position_embeddings"],
        hidden_act=ACTS[cfg["hidden_act"]],
        weight_type=weight_type,
        rope_theta=float(cfg.get("rope_theta", 10000)),
        head_dim=int(cfg.get("head_dim") or 0),
        norm_epsilon=float(cfg.get("rms_norm_eps", 1e-5)),
        n_experts=int(cfg.get("num_experts") or 0),
        n_active_experts=int(cfg.get("num_experts_per_tok") or 0),
        moe_hidden_dim=int(cfg.get("moe_intermediate_size") or 0),
    )
    rs = cfg.get("rope_scaling")
    if rs and rs.get("rope_type") == "llama3":
        h.rope_type = mf.ROPE_LLAMA3_1
        h.rope_scaling_factor = float(rs["factor"])
        h.rope_scaling_low_freq_factor = float(rs["low_freq_factor"])
        h.rope_scaling_high_freq_factor = float(rs["high_freq_factor"])
        h.rope_scaling_orig_max_seq_len = int(rs["original_max_position_embeddings"])
    h.finalize()
    return h


class HfTensors:
    """Lazy multi-file safetensors lookup."""

    def __init__(self, folder: str):
        from safetensors import safe_open
        self.files = [os.path.join(folder, f) for f in sorted(os.listdir(folder))
                      if f.endswith(".safetensors") and not f.startswith(".")]
        if not self.files:
            raise FileNotFoundError(f"no .safetensors in {folder}")
        self.index = {}
        self.handles = {}
        for path in self.files:
            h = safe_open(path, framework="np", device="cpu")
            self.handles[path] = h
            for key in h.keys():
                self.index[key] = path

    def get(self, *names):
        for name in names:
            if name in self.index:
                return np.asarray(self.handles[self.index[name]].get_tensor(name),
                                  dtype=np.float32)
        raise KeyError(f"tensor not found: {names}")


def convert(folder: str, weight_type: int, out_path: str) -> None:
    with open(os.path.join(folder, "config.json")) as f:
        cfg = json.load(f)
    h = header_from_config(cfg, weight_type)
    tensors = HfTensors(folder)
    arch_is_llama = h.arch_type == mf.ARCH_LLAMA
    wt = weight_type

    with open(out_path, "wb") as out:
        mf.write_header(out, h)

        def w(x, ftype):
            mf.write_tensor(out, x, ftype)

        w(tensors.get("model.embed_tokens.weight"), F32)
        for l in range(h.n_layers):
            pre = f"model.layers.{l}"
            q = tensors.get(f"{pre}.self_attn.q_proj.weight")
            k = tensors.get(f"{pre}.self_attn.k_proj.weight")
            if arch_is_llama:
                q = permute_qk(q, h.n_heads)
                k = permute_qk(k, h.n_kv_heads)
            w(q, wt)
            w(k, wt)
            w(tensors.get(f"{pre}.self_attn.v_proj.weight"), wt)
            w(tensors.get(f"{pre}.self_attn.o_proj.weight"), wt)
            if h.n_experts > 0:
                w(tensors.get(f"{pre}.mlp.gate.weight"), F32)
                for e in range(h.n_experts):
                    w(tensors.get(f"{pre}.mlp.experts.{e}.gate_proj.weight"), wt)
                    w(tensors.get(f"{pre}.mlp.experts.{e}.down_proj.weight"), wt)
                    w(tensors.get(f"{pre}.mlp.experts.{e}.up_proj.weight"), wt)
            else:
                w(tensors.get(f"{pre}.mlp.gate_proj.weight"), wt)
                w(tensors.get(f"{pre}.mlp.down_proj.weight"), wt)
                w(tensors.get(f"{pre}.mlp.up_proj.weight"), wt)
            if h.is_qwen3:
                w(tensors.get(f"{pre}.self_attn.q_norm.weight"), F32)
                w(tensors.get(f"{pre}.self_attn.k_norm.weight"), F32)
            w(tensors.get(f"{pre}.input_layernorm.weight"), F32)
            w(tensors.get(f"{pre}.post_attention_layernorm.weight"), F32)
        w(tensors.get("model.norm.weight"), F32)
        w(tensors.get("lm_head.weight", "model.embed_tokens.weight"), wt)
    print(f"✅ {out_path} created")


def main():
    if len(sys.argv) < 4:
        print(__doc__)
        return 1
    folder, ftype, name = sys.argv[1], sys.argv[2], sys.argv[3]
    convert(folder, FLOAT_TYPES[ftype], f"dllama_model_{name}_{ftype}.m")
    return 0


if __name__ == "__main__":
    sys.exit(main())
