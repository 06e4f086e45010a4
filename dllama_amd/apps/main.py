"""`dllama` CLI — inference / chat / perplexity / worker modes.

CLI parity with the reference (src/app.cpp:24-135, src/dllama.cpp): same
flags, same 🔶/🔷 per-run stats output (dllama.cpp:104-115). On MI355X the
reference's root+workers become torchrun ranks on one node:

  # 1 GPU
  dllama inference --model m.m --tokenizer t.t --prompt "Hello" --steps 64
  # 8 GPUs (replaces root + 7 TCP workers; reference README "How to run")
  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 -m dllama_amd.apps.main \
      inference --model m.m --tokenizer t.t --prompt "Hello" --steps 64
"""

from __future__ import annotations

import argparse
import os
import signal
import sys
import time

import numpy as np
import torch

from .. import model_file as mf
from ..engine import InferenceEngine
from ..models.config import ModelConfig
from ..parallel.comm import init_dist_comm
from ..quants import F32, Q80
from ..tokenizer import (ChatItem, ChatTemplateGenerator, EosDetector, Sampler,
                         TEMPLATE_UNKNOWN, Tokenizer, _TEMPLATE_NAMES, chat_stops)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="dllama")
    p.add_argument("mode", choices=["inference", "chat", "perplexity", "worker"])
    p.add_argument("--model", required=False)
    p.add_argument("--tokenizer", required=False)
    p.add_argument("--prompt", default=None)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--buffer-float-type", default="q80", choices=["q80", "f32"],
                   help="TP sync buffer quantization (reference --buffer-float-type)")
    p.add_argument("--nthreads", type=int, default=os.cpu_count(),
                   help="CPU backend threads (reference --nthreads)")
    p.add_argument("--cpu-dtype", default="f32", choices=["f32", "f16", "q40"],
                   help="CPU backend weight handling: q40 streams the Q40 "
                        "planes natively (quantized-weight RAM, native C++ "
                        "matmul — role of the reference AVX512/NEON path); "
                        "f16 halves resident memory via torch; f32 is the "
                        "exact-oracle default")
    p.add_argument("--temperature", type=float, default=0.8)
    p.add_argument("--topp", type=float, default=0.9)
    p.add_argument("--seed", type=int, default=None)
    p.add_argument("--chat-template", default=None,
                   choices=[None, *_TEMPLATE_NAMES.keys()])
    p.add_argument("--max-seq-len", type=int, default=0)
    p.add_argument("--gpu-index", type=int, default=None,
                   help="GPU to use (default LOCAL_RANK); -1 forces CPU")
    p.add_argument("--n-batches", type=int, default=32,
                   help="max prompt tokens per prefill step (reference nBatches)")
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph decode capture")
    p.add_argument("--show-timing", action="store_true",
                   help="per-token generation time (reference 🔶 lines)")
    # accepted for reference CLI parity; meaningless on one xGMI node
    p.add_argument("--workers", nargs="*", default=None,
                   help="ignored: TP ranks come from torchrun (xGMI, not TCP)")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=9990)
    p.add_argument("--net-turbo", type=int, default=1, help="ignored (no TCP mesh)")
    p.add_argument("--gpu-segments", default=None,
                   help="ignored: reference hybrid CPU/GPU device placement "
                        "(app.cpp:119-124); every rank here is one whole GPU")
    return p


def load_engine(args):
    """Model + tokenizer + engine with the right backend and TP setup."""
    comm = init_dist_comm()
    sync = Q80 if args.buffer_float_type == "q80" else F32
    m = mf.ModelFile(args.model, max_seq_len=args.max_seq_len, sync_type=sync)
    use_gpu = torch.cuda.is_available() and args.gpu_index != -1
    cfg = ModelConfig.from_header(m.header, world=comm.world, rank=comm.rank)
    if comm.rank == 0:
        _print_header(m.header, cfg, use_gpu)
    if use_gpu:
        from ..models.hip_model import HipTransformer
        dev = args.gpu_index if args.gpu_index is not None \
            else int(os.environ.get("LOCAL_RANK", "0"))
        model = HipTransformer.from_file(m, cfg, device=f"cuda:{dev}", comm=comm,
                                         n_batches=args.n_batches)
        if not args.no_graph:
            # RCCL collectives are hipGraph-capturable, so TP decode is one
            # graph replay per token too (every rank captures in lockstep —
            # the warmup forwards inside contain collectives)
            try:
                model.capture_decode_graph()
            except Exception as e:  # noqa: BLE001
                if comm.rank == 0:
                    print(f"⚠️  decode graph capture failed ({e}); "
                          "running eager (slower)", file=sys.stderr)
    else:
        from ..models.cpu_model import CpuTransformer
        torch.set_num_threads(max(1, args.nthreads))
        wdt = ("q40" if args.cpu_dtype == "q40"
               else torch.float16 if args.cpu_dtype == "f16" else torch.float32)
        model = CpuTransformer(m, cfg, comm, weight_dtype=wdt)
    tok = Tokenizer(args.tokenizer) if args.tokenizer else None
    seed = args.seed if args.seed is not None else int(time.time())
    if comm.world > 1:
        # all ranks must sample identically (lockstep decode): share rank 0's seed
        st = torch.tensor([seed], dtype=torch.int64)
        if torch.cuda.is_available():
            st = st.cuda()
        comm.broadcast_(st, src=0)
        seed = int(st.item())
    sampler = Sampler(m.header.vocab_size, args.temperature, args.topp, seed)
    nb = getattr(model, "n_batches", args.n_batches)  # HIP backend may round up
    return InferenceEngine(model, tok, sampler, n_batches=nb), m, comm


def _print_header(h, cfg, use_gpu):
    print(f"💡 Arch: {'Llama' if h.arch_type == mf.ARCH_LLAMA else 'Qwen3 MoE' if h.arch_type == mf.ARCH_QWEN3_MOE else 'Qwen3'}")
    print(f"💡 Dim: {h.dim}\n💡 HiddenDim: {h.hidden_dim}\n💡 nLayers: {h.n_layers}")
    print(f"💡 nHeads: {h.n_heads}\n💡 nKvHeads: {h.n_kv_heads}\n💡 HeadDim: {h.head_dim}")
    print(f"💡 VocabSize: {h.vocab_size}\n💡 SeqLen: {h.seq_len}")
    if h.n_experts:
        print(f"💡 nExperts: {h.n_experts}\n💡 nActiveExperts: {h.n_active_experts}")
    print(f"💡 Backend: {'MI355X HIP' if use_gpu else 'CPU'}  TP={cfg.world}")
    # memory accounting (reference 📀 prints, nn-core.cpp:175-189): per-rank
    # resident weight shard + dense f32 KV cache
    kv_bytes = 2 * cfg.n_layers * cfg.seq_len * cfg.kv_dim0 * 4
    weights = h.file_size - h.header_size  # .m = header + packed weights
    print(f"📀 Weights/rank: {weights / max(1, cfg.world) / 1e9:.2f} GB"
          f"  KV cache/rank: {kv_bytes / 1e9:.2f} GB (seq {cfg.seq_len})")


def run_inference(args) -> int:
    engine, m, comm = load_engine(args)
    if args.prompt is None:
        print("error: missing --prompt", file=sys.stderr)
        return 1
    tok = engine.tokenizer
    tokens = tok.encode(args.prompt) if tok else [int(t) for t in args.prompt.split()]
    quiet = comm.rank != 0

    pieces = []
    if tok:
        tok.reset_decoder()

    last = [time.perf_counter()]

    def on_token(t):
        if quiet or not tok:
            return
        if args.show_timing:
            now = time.perf_counter()
            print(f"🔶 P {1000 * (now - last[0]):6.2f} ms - {t}", flush=True)
            last[0] = now
            tok.decode(t)
            return
        piece = tok.decode(t)
        if piece:
            pieces.append(piece)
            print(piece, end="", flush=True)

    if not quiet:
        print(args.prompt, end="", flush=True)
    out, stats = engine.generate(tokens, args.steps, on_token=on_token,
                                 stop_check=(tok.is_eos if tok else None))
    if not quiet:
        print()
        # reference per-run stats format (dllama.cpp:104-115)
        print(f"Evaluation\n   nBatches: {args.n_batches}\n   nTokens: {stats.prefill_tokens}\n"
              f"   tokens/s: {stats.eval_tok_s:.2f} ({1000.0 / max(stats.eval_tok_s, 1e-9):.2f} ms/tok)")
        print(f"Prediction\n   nTokens: {stats.decode_tokens}\n"
              f"   tokens/s: {stats.pred_tok_s:.2f} ({1000.0 / max(stats.pred_tok_s, 1e-9):.2f} ms/tok)")
    return 0


def run_chat(args) -> int:
    """Interactive chat REPL (reference dllama.cpp:174-258)."""
    engine, m, comm = load_engine(args)
    tok = engine.tokenizer
    ttype = _TEMPLATE_NAMES.get(args.chat_template, TEMPLATE_UNKNOWN) \
        if args.chat_template else TEMPLATE_UNKNOWN
    eos_piece = tok.vocab[tok.eos_token_ids[0]].decode("utf-8", "replace") \
        if tok.eos_token_ids else ""
    gen = ChatTemplateGenerator(ttype, tok.chat_template, eos_piece)
    stops = chat_stops(tok)
    quiet = comm.rank != 0
    is_start = True
    # optional system prompt, included in the first turn's delta items
    # (reference dllama.cpp:182-185)
    try:
        sys_prompt = input("💻 System prompt (optional): ") if not quiet else input()
    except EOFError:
        return 0
    while True:
        try:
            user = input("\n👱 User\n> ") if not quiet else input()
        except EOFError:
            return 0
        if not user.strip():
            continue
        items = []
        if is_start and sys_prompt.strip():
            items.append(ChatItem("system", sys_prompt))
        items.append(ChatItem("user", user))
        chat_out = gen.generate(items, True)
        text = chat_out.content
        try:
            tokens = tok.encode(text, is_start=is_start)
        except ValueError:
            print("(encode error)", file=sys.stderr)
            continue
        is_start = False
        if engine.pos + len(tokens) + 1 >= m.header.seq_len:
            print("(end of context)")  # reference dllama.cpp:257
            return 0
        detector = EosDetector(tok.eos_token_ids, stops)
        tok.reset_decoder()
        if not quiet:
            # deepseek's "<think>\n" tail is part of the prompt but shown as
            # assistant output (reference publicPrompt, dllama.cpp:233-235)
            print("\n🤖 Assistant\n" + (chat_out.public_prompt or ""),
                  end="", flush=True)

        def on_token(t):
            # route through the streaming decoder first (UTF-8-safe pieces;
            # reference feeds tokenizer->decode output to the detector)
            piece = tok.decode(t)
            kind = detector.append(t, piece)
            if not quiet:
                delta = detector.get_delta()
                if delta and kind != 0:  # not MAYBE_EOS
                    print(delta, end="", flush=True)
                    detector.reset()

        engine.generate(tokens, args.steps, on_token=on_token,
                        stop_check=lambda t: detector.is_eos(t)
                        or detector.eos_pos >= 0)


def run_perplexity(args) -> int:
    """Perplexity over --prompt (reference dllama.cpp:132-172)."""
    engine, m, comm = load_engine(args)
    tok = engine.tokenizer
    tokens = tok.encode(args.prompt) if tok else [int(t) for t in args.prompt.split()]
    if len(tokens) < 2:
        print("error: need at least 2 tokens", file=sys.stderr)
        return 1
    nll, count = 0.0, 0
    pos = 0
    nb = engine.n_batches  # the backend may pin/round the CLI value
    for i in range(0, len(tokens) - 1, nb):
        chunk = tokens[i: i + nb]
        t = torch.tensor(chunk, dtype=torch.int64)
        p = torch.arange(pos, pos + len(chunk), dtype=torch.int64)
        logits = engine.model.forward(t, p)  # [B, vocab]
        logits = logits.detach().float().cpu()
        for j in range(len(chunk)):
            nxt_idx = i + j + 1
            if nxt_idx >= len(tokens):
                break
            row = logits[j]
            logp = row - torch.logsumexp(row, dim=-1)
            nll -= float(logp[tokens[nxt_idx]])
            count += 1
        pos += len(chunk)
    ppl = float(np.exp(nll / count))
    if comm.rank == 0:
        print(f"Perplexity: {ppl:.4f}  (nll/token {nll / count:.4f}, "
              f"bitPerToken {nll / count / np.log(2.0):.4f}, {count} tokens)")
    return 0


def main(argv=None) -> int:
    # die silently when stdout's reader goes away (e.g. `dllama ... | head`),
    # like any Unix CLI, instead of a BrokenPipeError traceback
    if hasattr(signal, "SIGPIPE"):
        signal.signal(signal.SIGPIPE, signal.SIG_DFL)
    args = build_parser().parse_args(argv)
    if args.mode == "worker":
        print("ℹ️  On MI355X, workers are torchrun ranks on one node — run:\n"
              "   torchrun --nproc-per-node N --master-addr 127.0.0.1 "
              "-m dllama_amd.apps.main inference ...\n"
              "   (replaces the reference's TCP root+worker mesh; "
              "non-zero ranks stay silent)")
        return 0
    if args.mode == "inference":
        return run_inference(args)
    if args.mode == "chat":
        return run_chat(args)
    if args.mode == "perplexity":
        return run_perplexity(args)
    return 1


if __name__ == "__main__":
    sys.exit(main())
