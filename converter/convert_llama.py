#!/usr/bin/env python3
"""Meta `consolidated.*.pth` checkpoint -> .m converter.

Role parity with reference converter/convert-llama.py: reads the original
Llama release format (params.json + consolidated shards, tensors already in
the interleaved-pair rope layout — no Q/K permute needed, unlike the HF
layout) and writes the canonical .m walk.

Usage: python converter/convert_llama.py <model_folder> <q40|q80|f32> <name>
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from dllama_amd import model_file as mf
from dllama_amd.quants import F32, Q40, Q80

FLOAT_TYPES = {"f32": F32, "q40": Q40, "q80": Q80}


def load_shards(folder: str):
    paths = sorted(p for p in os.listdir(folder)
                   if p.startswith("consolidated.") and p.endswith(".pth"))
    if not paths:
        raise FileNotFoundError(f"no consolidated.*.pth in {folder}")
    shards = [torch.load(os.path.join(folder, p), map_location="cpu",
                         weights_only=True) for p in paths]
    return shards


def gather(shards, key: str, dim: int | None):
    """Concatenate a tensor across Meta's column/row-parallel shards."""
    parts = [s[key] for s in shards if key in s]
    if not parts:
        raise KeyError(key)
    t = parts[0] if len(parts) == 1 or dim is None else torch.cat(parts, dim=dim)
    return t.to(torch.float32).numpy()


def convert(folder: str, weight_type: int, out_path: str) -> None:
    with open(os.path.join(folder, "params.json")) as f:
        params = json.load(f)
    shards = load_shards(folder)
    dim = params["dim"]
    n_heads = params["n_heads"]
    n_kv_heads = params.get("n_kv_heads", n_heads)
    n_layers = params["n_layers"]
    vocab = params.get("vocab_size", -1)
    if vocab <= 0:
        vocab = sum(s["tok_embeddings.weight"].shape[0] for s in shards
                    if "tok_embeddings.weight" in s)
    # Meta hidden dim derivation (multiple_of rounding)
    sample = gather(shards, "layers.0.feed_forward.w1.weight", 0)
    hidden_dim = sample.shape[0]

    h = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=dim, hidden_dim=hidden_dim,
                     n_layers=n_layers, n_heads=n_heads, n_kv_heads=n_kv_heads,
                     vocab_size=vocab, seq_len=params.get("max_seq_len", 2048),
                     rope_theta=float(params.get("rope_theta", 10000.0)),
                     weight_type=weight_type,
                     norm_epsilon=float(params.get("norm_eps", 1e-5)))
    h.finalize()

    with open(out_path, "wb") as out:
        mf.write_header(out, h)

        def w(x, ftype):
            mf.write_tensor(out, x, ftype)

        wt = weight_type
        w(gather(shards, "tok_embeddings.weight", 1), F32)
        for l in range(n_layers):
            pre = f"layers.{l}"
            w(gather(shards, f"{pre}.attention.wq.weight", 0), wt)
            w(gather(shards, f"{pre}.attention.wk.weight", 0), wt)
            w(gather(shards, f"{pre}.attention.wv.weight", 0), wt)
            w(gather(shards, f"{pre}.attention.wo.weight", 1), wt)
            w(gather(shards, f"{pre}.feed_forward.w1.weight", 0), wt)
            w(gather(shards, f"{pre}.feed_forward.w2.weight", 1), wt)
            w(gather(shards, f"{pre}.feed_forward.w3.weight", 0), wt)
            w(gather(shards, f"{pre}.attention_norm.weight", None), F32)
            w(gather(shards, f"{pre}.ffn_norm.weight", None), F32)
        w(gather(shards, "norm.weight", None), F32)
        w(gather(shards, "output.weight", 0), wt)
    print(f"✅ {out_path} created")


def main():
    if len(sys.argv) < 4:
        print(__doc__)
        return 1
    convert(sys.argv[1], FLOAT_TYPES[sys.argv[2]],
            f"dllama_model_{sys.argv[3]}_{sys.argv[2]}.m")
    return 0


if __name__ == "__main__":
    sys.exit(main())
