#!/usr/bin/env python3
"""Big-model file-load harness: prove the `.m` path at
real scale on one GPU.

Writes a real-size synthetic 70B (or other preset) `.m` with the fast tiled
writer, loads it through `HipTransformer.from_file` at TP=1, and decodes a
few tokens — measuring write, load (mmap walk + nibble-plane repack +
upload) and decode throughput. Exercises the exact code path BASELINE
configs 3/5 use on an 8-GPU node (reference weight walk: llm.cpp:614-669).

Run on a GPU box:
  python tools/load_bigmodel.py --model llama-3.3-70b --path /tmp/m70.m
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dllama_amd import model_file as mf  # noqa: E402
from dllama_amd.models.config import ModelConfig  # noqa: E402
from dllama_amd.models.hip_model import HipTransformer  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3.3-70b")
    ap.add_argument("--path", default="/tmp/dllama_big.m")
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--keep", action="store_true")
    args = ap.parse_args()

    h = mf.preset_header(args.model, seq_len=args.seq_len)
    t0 = time.time()
    mf.write_synthetic_model(args.path, h, fast=True)
    t_write = time.time() - t0
    size_gb = os.path.getsize(args.path) / 1e9
    print(f"write: {size_gb:.1f} GB in {t_write:.1f}s "
          f"({size_gb / t_write:.2f} GB/s)", flush=True)

    t0 = time.time()
    m = mf.ModelFile(args.path, max_seq_len=args.seq_len)
    cfg = ModelConfig.from_header(m.header)
    model = HipTransformer.from_file(m, cfg)
    torch.cuda.synchronize()
    t_load = time.time() - t0
    print(f"load (mmap walk + repack + upload): {t_load:.1f}s "
          f"({size_gb / t_load:.2f} GB/s)", flush=True)
    free, total = torch.cuda.mem_get_info()
    print(f"HBM used: {(total - free) / 1e9:.1f} GB of {total / 1e9:.0f} GB")

    model.greedy_feedback = True
    model.capture_decode_graph()
    model.pos.fill_(0)
    model.tokens[0] = 7
    for _ in range(5):
        model._graph.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        model._graph.replay()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"decode: {args.steps / dt:.1f} tok/s ({dt / args.steps * 1000:.2f} ms/tok) "
          f"TP=1 from-file weights")
    if not args.keep:
        os.remove(args.path)


if __name__ == "__main__":
    main()
