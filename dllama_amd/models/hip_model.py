"""MI355X transformer: device-resident Q40 weights + hand-written HIP kernels.

Architecture (MI355X-first):
  - weights repacked at load into the GEMV device layout (nibble plane
    uint8 [d, n/2] + f16 scale plane [d, n/32]) — the .m block stream is
    only a wire format;
  - all activations live in preallocated device buffers sized for the max
    batch, so the whole decode step is hipGraph-capturable (reference
    replays recorded Vulkan command buffers the same way,
    nn-vulkan.cpp:1065-1118);
  - the position is a device int32 tensor read by rope/kv/attention kernels
    — no host round-trip per token;
  - TP sync = on-device Q80 pack -> RCCL all-gather -> merge-add kernel
    (reference SYNC_NODE_SLICES + OP_MERGE_ADD) or plain f32 all-reduce
    (sync_type f32).
"""

from __future__ import annotations

import numpy as np
import torch

from .. import model_file as mf
from ..model_file import HIDDEN_ACT_GELU, ModelFile, ROPE_FALCON
from ..ops import hip_ops
from ..ops import reference as R
from ..parallel.comm import Comm, SingleComm
from ..quants import Q80, q40_to_planes
from .config import ModelConfig

QB = 32


class Linear:
    """A Q40 linear layer shard on device: y = W x."""

    def __init__(self, qs: torch.Tensor, scales: torch.Tensor):
        self.qs = qs          # uint8 [d, n/2] (or [E, d, n/2])
        self.scales = scales  # f16  [d, n/32] (or [E, d, n/32])
        self.d = qs.shape[-2]
        self.n = qs.shape[-1] * 2

    @classmethod
    def from_blocks(cls, raw: np.ndarray, d: int, n: int, device) -> "Linear":
        qs, sc = q40_to_planes(raw, d, n)
        return cls(torch.from_numpy(qs).to(device),
                   torch.from_numpy(sc).to(device))

    @classmethod
    def synthetic(cls, d: int, n: int, device, gen: torch.Generator,
                  scale: float = 0.02) -> "Linear":
        qs = torch.randint(0, 256, (d, n // 2), dtype=torch.uint8,
                           device=device, generator=gen)
        sc = (torch.rand((d, n // QB), device=device, generator=gen) * scale / 8)
        return cls(qs, sc.to(torch.float16))


class QuantBuf:
    """Q80 activation triple for a [rows, n] buffer."""

    def __init__(self, rows: int, n: int, device):
        self.n = n
        self.q = torch.zeros(rows, n, dtype=torch.int8, device=device)
        self.s = torch.zeros(rows, n // QB, dtype=torch.float32, device=device)
        self.bs = torch.zeros(rows, n // QB, dtype=torch.float32, device=device)


def _pow2_batch(b: int) -> int:
    nb = 1
    while nb < b:
        nb *= 2
    return nb


class HipTransformer:
    def __init__(self, config: ModelConfig, device=None, comm: Comm | None = None,
                 n_batches: int = 32):
        self.cfg = config
        self.comm = comm or SingleComm()
        self.device = torch.device(device or "cuda")
        self.k = hip_ops()
        self.n_batches = n_batches
        self.layers: list[dict] = []
        self.embedding = None
        self.final_norm = None
        self.wcls = None
        self._graph = None
        self._graph_pos = None
        self.greedy_feedback = False
        self._alloc_buffers()

    # ------------------------------------------------------------ weights

    @classmethod
    def from_file(cls, m: ModelFile, config: ModelConfig, device=None,
                  comm: Comm | None = None, n_batches: int = 32) -> "HipTransformer":
        self = cls(config, device, comm, n_batches)
        c, dev = config, self.device
        r, w = c.rank, c.world

        def lin(name, layer, d, n, expert=-1):
            e = m.entry(name, layer, expert)
            raw = m.slice_bytes(e, r, w)
            return Linear.from_blocks(raw, d, n, dev)

        def f32(name, layer=-1):
            return torch.from_numpy(np.array(m.f32(name, layer))).to(dev)

        self.embedding = f32("embedding")
        self.final_norm = f32("final_norm")
        self.wcls = lin("final_matmul_logits", -1, c.vocab0, c.dim)
        for l in range(c.n_layers):
            lw = {
                "q": lin("block_matmul_q", l, c.q_dim0, c.dim),
                "k": lin("block_matmul_k", l, c.kv_dim0, c.dim),
                "v": lin("block_matmul_v", l, c.kv_dim0, c.dim),
                "wo": lin("block_matmul_wo", l, c.dim, c.q_dim0),
                "norm0": f32("block_norm_0", l),
                "norm1": f32("block_norm_1", l),
            }
            if c.is_moe:
                lw["gate"] = f32("block_moe_gate", l)
                for wn, dd, nn in (("w1", c.ff_dim0, c.dim), ("w2", c.dim, c.ff_dim0),
                                   ("w3", c.ff_dim0, c.dim)):
                    ls = [lin(f"block_matmul_{wn}", l, dd, nn, e)
                          for e in range(c.n_experts)]
                    lw[wn] = Linear(torch.stack([x.qs for x in ls]),
                                    torch.stack([x.scales for x in ls]))
            else:
                lw["w1"] = lin("block_matmul_w1", l, c.ff_dim0, c.dim)
                lw["w2"] = lin("block_matmul_w2", l, c.dim, c.ff_dim0)
                lw["w3"] = lin("block_matmul_w3", l, c.ff_dim0, c.dim)
            if c.is_qwen3:
                lw["q_norm"] = f32("block_norm_q", l)
                lw["k_norm"] = f32("block_norm_k", l)
            self.layers.append(lw)
        self._finish_init()
        return self

    @classmethod
    def synthetic(cls, config: ModelConfig, device=None, comm: Comm | None = None,
                  n_batches: int = 32, seed: int = 1234) -> "HipTransformer":
        """Random-init weights built directly on device (benches: no network
        for checkpoints, and an 8B .m file round-trip is pointless there)."""
        self = cls(config, device, comm, n_batches)
        c, dev = config, self.device
        gen = torch.Generator(device=dev)
        gen.manual_seed(seed + c.rank)
        self.embedding = torch.randn(c.vocab_size, c.dim, device=dev,
                                     generator=gen) * 0.02
        self.final_norm = torch.ones(c.dim, device=dev)
        self.wcls = Linear.synthetic(c.vocab0, c.dim, dev, gen)
        for l in range(c.n_layers):
            lw = {
                "q": Linear.synthetic(c.q_dim0, c.dim, dev, gen),
                "k": Linear.synthetic(c.kv_dim0, c.dim, dev, gen),
                "v": Linear.synthetic(c.kv_dim0, c.dim, dev, gen),
                "wo": Linear.synthetic(c.dim, c.q_dim0, dev, gen),
                "norm0": torch.ones(c.dim, device=dev),
                "norm1": torch.ones(c.dim, device=dev),
            }
            if c.is_moe:
                lw["gate"] = torch.randn(c.n_experts, c.dim, device=dev,
                                         generator=gen) * 0.02
                for wn, dd, nn in (("w1", c.ff_dim0, c.dim), ("w2", c.dim, c.ff_dim0),
                                   ("w3", c.ff_dim0, c.dim)):
                    qs = torch.randint(0, 256, (c.n_experts, dd, nn // 2),
                                       dtype=torch.uint8, device=dev, generator=gen)
                    sc = (torch.rand((c.n_experts, dd, nn // QB), device=dev,
                                     generator=gen) * 0.02 / 8).to(torch.float16)
                    lw[wn] = Linear(qs, sc)
            else:
                lw["w1"] = Linear.synthetic(c.ff_dim0, c.dim, dev, gen)
                lw["w2"] = Linear.synthetic(c.dim, c.ff_dim0, dev, gen)
                lw["w3"] = Linear.synthetic(c.ff_dim0, c.dim, dev, gen)
            if c.is_qwen3:
                lw["q_norm"] = torch.ones(c.head_dim, device=dev)
                lw["k_norm"] = torch.ones(c.head_dim, device=dev)
            self.layers.append(lw)
        self._finish_init()
        return self

    def _alloc_buffers(self):
        c, dev, NB = self.cfg, self.device, self.n_batches
        self.pos = torch.zeros(1, dtype=torch.int32, device=dev)
        self.tokens = torch.zeros(NB, dtype=torch.int64, device=dev)
        self.x = torch.zeros(NB, c.dim, device=dev)
        self.t_norm = torch.zeros(NB, c.dim, device=dev)
        self.xq = QuantBuf(NB, c.dim, dev)
        self.qbuf = torch.zeros(NB, c.q_dim0, device=dev)
        self.kbuf = torch.zeros(NB, c.kv_dim0, device=dev)
        self.vbuf = torch.zeros(NB, c.kv_dim0, device=dev)
        self.zbuf = torch.zeros(NB, c.q_dim0, device=dev)
        self.zq = QuantBuf(NB, c.q_dim0, dev)
        self.partial = torch.zeros(NB, c.dim, device=dev)
        ffw = c.ff_dim0
        self.abuf = torch.zeros(NB, ffw, device=dev)
        self.gbuf = torch.zeros(NB, ffw, device=dev)
        self.dq = QuantBuf(NB, ffw, dev)
        self.logits0 = torch.zeros(NB, c.vocab0, device=dev)
        if c.world > 1:
            self.logits_gather = torch.zeros(c.world, NB, c.vocab0, device=dev)
            if c.sync_type == Q80:
                row_bytes = c.dim + 2 * (c.dim // QB)
                self.sync_out = torch.zeros(NB * row_bytes, dtype=torch.uint8, device=dev)
                self.sync_in = torch.zeros(c.world, NB * row_bytes,
                                           dtype=torch.uint8, device=dev)
        if c.is_moe:
            S = NB * c.n_active_experts
            self.moe_idx = torch.zeros(S, dtype=torch.int32, device=dev)
            self.moe_w = torch.zeros(NB, c.n_active_experts, device=dev)
            self.moe_a = torch.zeros(S, c.ff_dim0, device=dev)
            self.moe_g = torch.zeros(S, c.ff_dim0, device=dev)
            self.moe_dq = QuantBuf(S, c.ff_dim0, dev)
            self.moe_y = torch.zeros(S, c.dim, device=dev)

    def _finish_init(self):
        c, dev = self.cfg, self.device
        cache = R.rope_cache(c.seq_len, c.head_dim, c.rope_theta, c.rope_scaling)
        self.rope_cache = cache.reshape(c.seq_len, c.head_dim).contiguous().to(dev)
        self.k_cache = [torch.zeros(c.seq_len, c.kv_dim0, device=dev)
                        for _ in range(c.n_layers)]
        self.v_cache = [torch.zeros(c.seq_len, c.kv_dim0, device=dev)
                        for _ in range(c.n_layers)]
        self.rope_style = 1 if c.rope_type == ROPE_FALCON else 0

    # ------------------------------------------------------------ forward

    def _sync_partial(self, B: int, NB: int):
        """All-reduce self.partial[:NB] across ranks into x += sum(partials)."""
        c = self.cfg
        if c.world == 1:
            self.k.add_(self.x[:NB], self.partial[:NB])
            return
        if c.sync_type == Q80:
            nb_dim = c.dim // QB
            q = self.xq  # reuse dim-sized quant buffer
            self.k.q80_quantize(self.partial[:NB], q.q[:NB], q.s[:NB], q.bs[:NB])
            row_bytes = c.dim + 2 * nb_dim
            out = self.sync_out[: NB * row_bytes]
            self.k.sync_pack(q.q[:NB], q.s[:NB], out)
            inb = self.sync_in[:, : NB * row_bytes]
            self.comm.all_gather(inb, out)
            self.k.merge_add(self.x[:NB], inb)
        else:
            self.comm.allreduce_(self.partial[:NB])
            self.k.add_(self.x[:NB], self.partial[:NB])

    def forward_buffers(self, B: int):
        """Run one step over tokens[:B] at positions pos..pos+B-1, writing
        logits into logits0 (and the gather buffer under TP). Everything
        stays on device — this function is graph-capturable."""
        c, k = self.cfg, self.k
        NB = _pow2_batch(B)
        x = self.x
        torch.index_select(self.embedding, 0, self.tokens[:NB], out=x[:NB])

        kv_mul = c.n_heads0 // max(1, c.kv_dim0 // c.head_dim)
        for l, lw in enumerate(self.layers):
            # attention block
            k.rmsnorm_q80(x[:NB], lw["norm0"], self.xq.q[:NB], self.xq.s[:NB],
                          self.xq.bs[:NB], c.norm_eps)
            k.q40_gemv(lw["q"].qs, lw["q"].scales, self.xq.q, self.xq.s,
                       self.xq.bs, self.qbuf, NB)
            k.q40_gemv(lw["k"].qs, lw["k"].scales, self.xq.q, self.xq.s,
                       self.xq.bs, self.kbuf, NB)
            k.q40_gemv(lw["v"].qs, lw["v"].scales, self.xq.q, self.xq.s,
                       self.xq.bs, self.vbuf, NB)
            if c.is_qwen3:
                k.rmsnorm_rows(self.qbuf[:B].view(-1, c.head_dim), lw["q_norm"],
                               self.qbuf[:B].view(-1, c.head_dim), c.norm_eps)
                k.rmsnorm_rows(self.kbuf[:B].view(-1, c.head_dim), lw["k_norm"],
                               self.kbuf[:B].view(-1, c.head_dim), c.norm_eps)
            k.rope(self.qbuf[:B], self.rope_cache, self.pos, c.head_dim, self.rope_style)
            k.rope(self.kbuf[:B], self.rope_cache, self.pos, c.head_dim, self.rope_style)
            k.kv_append(self.kbuf[:B], self.vbuf[:B], self.k_cache[l],
                        self.v_cache[l], self.pos)
            k.attn(self.qbuf[:B], self.k_cache[l], self.v_cache[l], self.zbuf[:B],
                   self.pos, B, c.n_heads0, kv_mul, c.head_dim)
            k.q80_quantize(self.zbuf[:NB], self.zq.q[:NB], self.zq.s[:NB],
                           self.zq.bs[:NB])
            k.q40_gemv(lw["wo"].qs, lw["wo"].scales, self.zq.q, self.zq.s,
                       self.zq.bs, self.partial, NB)
            self._sync_partial(B, NB)

            # ffn block
            k.rmsnorm(x[:NB], lw["norm1"], self.t_norm[:NB], c.norm_eps)
            if c.is_moe:
                self._moe_ffn(B, NB, lw)
            else:
                k.q80_quantize(self.t_norm[:NB], self.xq.q[:NB], self.xq.s[:NB],
                               self.xq.bs[:NB])
                k.q40_gemv(lw["w1"].qs, lw["w1"].scales, self.xq.q, self.xq.s,
                           self.xq.bs, self.abuf, NB)
                k.q40_gemv(lw["w3"].qs, lw["w3"].scales, self.xq.q, self.xq.s,
                           self.xq.bs, self.gbuf, NB)
                k.swiglu_q80(self.abuf[:NB], self.gbuf[:NB], self.dq.q[:NB],
                             self.dq.s[:NB], self.dq.bs[:NB])
                k.q40_gemv(lw["w2"].qs, lw["w2"].scales, self.dq.q, self.dq.s,
                           self.dq.bs, self.partial, NB)
            self._sync_partial(B, NB)

        k.rmsnorm_q80(x[:NB], self.final_norm, self.xq.q[:NB], self.xq.s[:NB],
                      self.xq.bs[:NB], c.norm_eps)
        k.q40_gemv(self.wcls.qs, self.wcls.scales, self.xq.q, self.xq.s,
                   self.xq.bs, self.logits0, NB)
        if c.world > 1:
            self.comm.all_gather(self.logits_gather[:, :NB], self.logits0[:NB])
        if self.greedy_feedback and B == 1:
            # on-device greedy sampling feeding the next decode step (used by
            # the fully graph-captured bench loop; real serving samples on host)
            if c.world > 1:
                full = self.logits_gather[:, 0].reshape(-1)
                self.tokens[0].copy_(torch.argmax(full))
            else:
                self.tokens[0].copy_(torch.argmax(self.logits0[0]))

    def _moe_ffn(self, B: int, NB: int, lw: dict):
        """Router (torch) + grouped expert GEMVs (reference llm.cpp:450-487)."""
        c, k = self.cfg, self.k
        ka = c.n_active_experts
        router = self.t_norm[:NB] @ lw["gate"].t()
        probs = torch.softmax(router.float(), dim=-1)
        wts, idx = torch.topk(probs, ka, dim=-1)
        wts = wts / wts.sum(dim=-1, keepdim=True)
        self.moe_idx[: NB * ka].copy_(idx.reshape(-1).to(torch.int32))
        S = NB * ka
        k.q80_quantize(self.t_norm[:NB], self.xq.q[:NB], self.xq.s[:NB],
                       self.xq.bs[:NB])
        k.q40_gemv_grouped(lw["w1"].qs, lw["w1"].scales, self.xq.q, self.xq.s,
                           self.xq.bs, self.moe_idx[:S], self.moe_a, ka)
        k.q40_gemv_grouped(lw["w3"].qs, lw["w3"].scales, self.xq.q, self.xq.s,
                           self.xq.bs, self.moe_idx[:S], self.moe_g, ka)
        k.swiglu_q80(self.moe_a[:S], self.moe_g[:S], self.moe_dq.q[:S],
                     self.moe_dq.s[:S], self.moe_dq.bs[:S])
        k.q40_gemv_grouped(lw["w2"].qs, lw["w2"].scales, self.moe_dq.q,
                           self.moe_dq.s, self.moe_dq.bs, self.moe_idx[:S],
                           self.moe_y, 1)
        y = self.moe_y[:S].reshape(NB, ka, c.dim)
        torch.sum(y * wts.unsqueeze(-1), dim=1, out=self.partial[:NB])

    # ------------------------------------------------------------ engine API

    def forward(self, tokens: torch.Tensor, positions: torch.Tensor) -> torch.Tensor:
        """Engine-compatible forward. positions must be a contiguous range.

        When a decode graph is captured and B==1, this is a single hipGraph
        replay (the graph advances the device position itself)."""
        B = tokens.shape[0]
        assert B <= self.n_batches
        p0 = int(positions[0])
        self.tokens[:B].copy_(tokens.to(self.device), non_blocking=True)
        if B == 1 and self._graph is not None:
            if p0 != self._graph_pos:
                self.pos.fill_(p0)
            self._graph.replay()
            self._graph_pos = p0 + 1  # the graph's pos_inc advanced it
        else:
            self.pos.fill_(p0)
            self._graph_pos = None
            self.forward_buffers(B)
        c = self.cfg
        if c.world > 1:
            return (self.logits_gather[:, :B].permute(1, 0, 2)
                    .reshape(B, c.vocab_size))
        return self.logits0[:B]

    # ------------------------------------------------------------ graphs

    def capture_decode_graph(self):
        """Capture the whole B=1 decode step (forward + pos advance) as a
        hipGraph; each subsequent decode costs one graph replay
        (the HIP analog of the reference's recorded Vulkan command buffers,
        nn-vulkan.cpp:1065-1118)."""
        torch.cuda.synchronize(self.device)
        s = torch.cuda.Stream(self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(2):  # warmup allocations/kernels on a side stream
                self.forward_buffers(1)
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self.forward_buffers(1)
            self.k.pos_inc(self.pos, 1)
        self._graph = g
        self._graph_pos = None  # device pos unknown until first fill
