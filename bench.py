#!/usr/bin/env python3
"""Flagship benchmark: Llama-3.1-8B Q40 single-token decode on MI355X.

Metric of record (BASELINE.json): decode ms/token -> tokens/sec, measured
at TP=1/2/4/8 on one node. Synthetic data + random-init weights of the real
architecture (no network for checkpoints). The whole decode step — fused
rmsnorm+quant, Q40 GEMVs, rope, KV append, flash attention, SwiGLU, TP sync,
logits, greedy sampling — runs per step; nothing is cached or skipped.

vs_baseline compares against the reference's published table
(report.pdf Figs. 3-6): the closest published config is
Llama 2 7B Q40 on Raspberry Pi 4B clusters (the reference publishes no
Llama-3.1-8B numbers); ratios use the same device count.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  (N>1 via torch.distributed.run, one rank per GPU over RCCL)
"""

import argparse
import json
import os
import sys
import time

import torch


# reference published decode throughput tok/s by device count
# (Llama 2 7B Q40 total ms/token on 1/2/4/8 RPi 4B)
BASELINE_TOKS = {1: 1000.0 / 1312.50, 2: 1000.0 / 793.69,
                 4: 1000.0 / 494.00, 8: 1000.0 / 588.19}


def _build_cpu_smoke(args, world, comm):
    """Tiny CPU model wired through the SAME TP plumbing (torchrun env,
    comm collectives, sync type) — used by tests to validate the driver's
    N>1 launch contract on machines without a GPU."""
    import tempfile

    from dllama_amd import model_file as mf
    from dllama_amd.models.config import ModelConfig
    from dllama_amd.models.cpu_model import CpuTransformer
    from dllama_amd.quants import F32, Q80

    path = os.path.join(tempfile.gettempdir(),
                        f"dllama_bench_smoke_{os.getppid()}.m")
    if comm.rank == 0 and not os.path.exists(path):
        # 8 kv heads so the smoke works at any driver TP degree (1..8)
        h = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=64, hidden_dim=128,
                         n_layers=2, n_heads=8, n_kv_heads=8, head_dim=64,
                         vocab_size=256, seq_len=128, rope_theta=10000,
                         rope_type=mf.ROPE_LLAMA)
        h.finalize()
        mf.write_synthetic_model(path, h, seed=5)
    comm.barrier()
    sync = Q80 if args.sync == "q80" else F32
    m = mf.ModelFile(path, sync_type=sync)
    cfg = ModelConfig.from_header(m.header, world=world, rank=comm.rank)
    cfg.sync_type = sync
    return CpuTransformer(m, cfg, comm)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--model", default="llama-3.1-8b")
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--prefill", "--ctx", type=int, default=32, dest="prefill",
                    help="prompt tokens evaluated before the timed decode "
                         "(--ctx 1024 measures long-context decode)")
    ap.add_argument("--no-graph", action="store_true")
    ap.add_argument("--sync", choices=["q80", "f32"], default="q80")
    # hidden: --device cpu runs a tiny model through the SAME distributed
    # plumbing (torchrun env, init_dist_comm, barriers, max-over-ranks,
    # rank-0 JSON) so tests can validate the driver contract without a GPU
    ap.add_argument("--device", choices=["cuda", "cpu"], default="cuda",
                    help=argparse.SUPPRESS)
    args = ap.parse_args()
    use_cpu = args.device == "cpu"

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from dllama_amd import model_file as mf
    from dllama_amd.models.config import ModelConfig
    from dllama_amd.models.hip_model import HipTransformer
    from dllama_amd.parallel.comm import init_dist_comm, SingleComm
    from dllama_amd.quants import F32, Q80

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.gpus > 1 and world == 1:
        print("error: --gpus N>1 must be launched via torch.distributed.run",
              file=sys.stderr)
        sys.exit(2)
    n_gpus = world if world > 1 else 1
    comm = init_dist_comm() if world > 1 else SingleComm()
    rank = comm.rank
    if use_cpu:
        device = torch.device("cpu")
    else:
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0")))
        torch.cuda.set_device(device)

    def sync_device():
        if not use_cpu:
            torch.cuda.synchronize(device)

    t0 = time.time()
    if use_cpu:
        model = _build_cpu_smoke(args, n_gpus, comm)
    else:
        header = mf.preset_header(args.model, seq_len=args.seq_len)
        header.sync_type = Q80 if args.sync == "q80" else F32
        cfg = ModelConfig.from_header(header, world=n_gpus, rank=rank)
        model = HipTransformer.synthetic(cfg, device=device, comm=comm)
        model.greedy_feedback = True
    cfg = model.cfg
    sync_device()
    if rank == 0:
        built = "tiny-llama-cpu-smoke" if use_cpu else f"synthetic {args.model}"
        print(f"# built {built} TP={n_gpus} in {time.time()-t0:.1f}s",
              file=sys.stderr)

    # short prefill so the decode attends over a non-trivial context
    # (seeded so every TP rank feeds identical tokens)
    torch.manual_seed(1234)
    prompt = torch.randint(0, cfg.vocab_size, (args.prefill,))
    for i in range(0, args.prefill, 32):
        chunk = prompt[i: i + 32]
        model.skip_logits = i + 32 < args.prefill
        model.forward(chunk, torch.arange(i, i + len(chunk)))
        model.skip_logits = False

    use_graph = not args.no_graph and not use_cpu
    if use_graph:
        try:
            # pick the K-split count for the timed region's context length
            # (the graph replays below bypass forward()'s adaptive recapture)
            sp = model._pick_splits(args.prefill)
            if sp != model.attn_splits:
                model._set_attn_splits(sp)
            model.capture_decode_graph()
            model.pos.fill_(args.prefill)
            model._graph_pos = args.prefill
        except Exception as e:  # noqa: BLE001
            if rank == 0:
                print(f"# graph capture failed ({e}); running eager", file=sys.stderr)
            use_graph = False

    if use_cpu:
        pos_h = [args.prefill]
        tok_t = torch.tensor([7])

        def step():
            model.forward(tok_t, torch.tensor([pos_h[0]]))
            pos_h[0] += 1
    else:
        def step():
            if use_graph:
                model._graph.replay()
            else:
                # eager decode: same kernels, per-op launches
                model.forward_buffers(1)
                model.k.pos_inc(model.pos, 1)

        if not use_graph:
            model.pos.fill_(args.prefill)
        model.tokens[0] = 7

    for _ in range(args.warmup):
        step()
    comm.barrier()
    sync_device()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync_device()
    elapsed = time.perf_counter() - t0
    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    comm.barrier()

    ms_per_step = elapsed / args.steps * 1000.0
    toks = args.steps / elapsed
    base = BASELINE_TOKS.get(n_gpus)
    if rank == 0:
        mname = "Llama-3.1-8B" if args.model == "llama-3.1-8b" else args.model
        print(json.dumps({
            "metric": f"decode tokens/s ({mname} Q40)",
            "value": round(toks, 2),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": round(toks / base, 1) if base and not use_cpu else None,
            "dtype": "f32-accum/int8-dot (Q40 weights, Q80 activations)",
            "data": "synthetic (random-init weights, random prompt; no network for checkpoints)",
            "config": {
                "model": args.model if not use_cpu else "tiny-llama-cpu-smoke",
                "global_batch": 1,
                "seq_len": args.seq_len,
                "prefill": args.prefill,
                "parallelism": f"tp{n_gpus}",
                "sync": args.sync,
                "graph": use_graph,
                "baseline_note": "vs Llama-2-7B-Q40 on N RPi-4B (reference report.pdf Fig.3; no 8B number published)",
            },
        }))


if __name__ == "__main__":
    main()
