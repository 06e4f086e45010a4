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
                 n_batches: int = 32, force_sync: bool = False):
        self.cfg = config
        self.comm = comm or SingleComm()
        # force_sync: run the full TP sync/gather/concat path at world=1
        # (SingleComm collectives are identity) — lets a 1-GPU box validate
        # every TP kernel and the graph-captured sync step end to end
        self.tp_path = config.world > 1 or force_sync
        self.device = torch.device(device or "cuda")
        self.k = hip_ops()
        # buffers are indexed with NB=_pow2_batch(B) rows and the prefill
        # GEMM's MFMA fragment covers exactly 32 batch rows: a non-pow2 or
        # <32 n_batches would launch grids over short buffers (OOB), and
        # >32 rows would silently never be written by the GEMM. Pin to 32.
        nb = 32
        if n_batches != 32:
            import warnings
            warnings.warn(
                f"HIP backend pins --n-batches to 32 (got {n_batches}): the "
                "prefill GEMM's MFMA fragment is 32 batch rows")
        self.n_batches = nb
        self.layers: list[dict] = []
        self.embedding = None
        self.final_norm = None
        self.wcls = None
        self._graph = None
        self._graph_pos = None
        self.greedy_feedback = False
        self.skip_logits = False  # prefill chunks before the last skip wcls
        self._pf_graphs = {}      # skip_logits -> captured 32-token graph
        self._pf_failed = False
        self._alloc_buffers()

    # ------------------------------------------------------------ weights

    @classmethod
    def from_file(cls, m: ModelFile, config: ModelConfig, device=None,
                  comm: Comm | None = None, n_batches: int = 32,
                  force_sync: bool = False) -> "HipTransformer":
        from ..quants import Q40
        if m.header.weight_type != Q40:
            raise ValueError(
                "the MI355X HIP backend runs Q40 weights (the reference's "
                "shipped format); f32/q80 .m files run on the CPU backend "
                "(--gpu-index -1) or can be re-quantized with convert_hf.py")
        self = cls(config, device, comm, n_batches, force_sync)
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
            # fuse Q|K|V (and W1|W3) into single GEMV weights: one launch,
            # one pass over x, full-chip grids even for the small K/V shards
            q = lin("block_matmul_q", l, c.q_dim0, c.dim)
            kk = lin("block_matmul_k", l, c.kv_dim0, c.dim)
            v = lin("block_matmul_v", l, c.kv_dim0, c.dim)
            lw = {
                "qkv": Linear(torch.cat([q.qs, kk.qs, v.qs]),
                              torch.cat([q.scales, kk.scales, v.scales])),
                "wo": lin("block_matmul_wo", l, c.dim, c.q_dim0),
                "norm0": f32("block_norm_0", l),
                "norm1": f32("block_norm_1", l),
            }
            if c.is_moe:
                lw["gate"] = f32("block_moe_gate", l)
                w13 = []
                for e in range(c.n_experts):
                    w1 = lin("block_matmul_w1", l, c.ff_dim0, c.dim, e)
                    w3 = lin("block_matmul_w3", l, c.ff_dim0, c.dim, e)
                    w13.append(Linear(torch.cat([w1.qs, w3.qs]),
                                      torch.cat([w1.scales, w3.scales])))
                lw["w13"] = Linear(torch.stack([x.qs for x in w13]),
                                   torch.stack([x.scales for x in w13]))
                w2 = [lin("block_matmul_w2", l, c.dim, c.ff_dim0, e)
                      for e in range(c.n_experts)]
                lw["w2"] = Linear(torch.stack([x.qs for x in w2]),
                                  torch.stack([x.scales for x in w2]))
            else:
                w1 = lin("block_matmul_w1", l, c.ff_dim0, c.dim)
                w3 = lin("block_matmul_w3", l, c.ff_dim0, c.dim)
                lw["w13"] = Linear(torch.cat([w1.qs, w3.qs]),
                                   torch.cat([w1.scales, w3.scales]))
                lw["w2"] = lin("block_matmul_w2", l, c.dim, c.ff_dim0)
            if c.is_qwen3:
                lw["q_norm"] = f32("block_norm_q", l)
                lw["k_norm"] = f32("block_norm_k", l)
            self.layers.append(lw)
        self._finish_init()
        return self

    @classmethod
    def synthetic(cls, config: ModelConfig, device=None, comm: Comm | None = None,
                  n_batches: int = 32, seed: int = 1234,
                  force_sync: bool = False) -> "HipTransformer":
        """Random-init weights built directly on device (benches: no network
        for checkpoints, and an 8B .m file round-trip is pointless there)."""
        self = cls(config, device, comm, n_batches, force_sync)
        c, dev = config, self.device
        gen = torch.Generator(device=dev)
        gen.manual_seed(seed + c.rank)
        self.embedding = torch.randn(c.vocab_size, c.dim, device=dev,
                                     generator=gen) * 0.02
        self.final_norm = torch.ones(c.dim, device=dev)
        self.wcls = Linear.synthetic(c.vocab0, c.dim, dev, gen)
        for l in range(c.n_layers):
            lw = {
                "qkv": Linear.synthetic(c.q_dim0 + 2 * c.kv_dim0, c.dim, dev, gen),
                "wo": Linear.synthetic(c.dim, c.q_dim0, dev, gen),
                "norm0": torch.ones(c.dim, device=dev),
                "norm1": torch.ones(c.dim, device=dev),
            }
            if c.is_moe:
                lw["gate"] = torch.randn(c.n_experts, c.dim, device=dev,
                                         generator=gen) * 0.02
                for wn, dd, nn in (("w13", 2 * c.ff_dim0, c.dim),
                                   ("w2", c.dim, c.ff_dim0)):
                    qs = torch.randint(0, 256, (c.n_experts, dd, nn // 2),
                                       dtype=torch.uint8, device=dev, generator=gen)
                    sc = (torch.rand((c.n_experts, dd, nn // QB), device=dev,
                                     generator=gen) * 0.02 / 8).to(torch.float16)
                    lw[wn] = Linear(qs, sc)
            else:
                lw["w13"] = Linear.synthetic(2 * c.ff_dim0, c.dim, dev, gen)
                lw["w2"] = Linear.synthetic(c.dim, c.ff_dim0, dev, gen)
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
        self.qkv_ld = c.q_dim0 + 2 * c.kv_dim0
        self.qkv_out = torch.zeros(NB, self.qkv_ld, device=dev)
        self.zbuf = torch.zeros(NB, c.q_dim0, device=dev)
        self.zq = QuantBuf(NB, c.q_dim0, dev)
        self.partial = torch.zeros(NB, c.dim, device=dev)
        self.ff_out = torch.zeros(NB, 2 * c.ff_dim0, device=dev)
        self.dq = QuantBuf(NB, c.ff_dim0, dev)
        self.logits0 = torch.zeros(NB, c.vocab0, device=dev)
        rpw = 2 if c.vocab0 >= 2048 else 1
        self.amax_blocks = -(-c.vocab0 // (4 * rpw))  # mirrors gemv RPW choice
        self.gemm_part = torch.zeros(32 * 32 * 8192, device=dev)  # K-split partials
        # sum-of-squares accumulators: [slot, batch, 16 spread x 32 pad]
        # (16 slots each on their own cacheline; atomics to one line serialize)
        self.ssq = torch.zeros(2 * c.n_layers + 1, NB, 16 * 32, device=dev)
        self.amax_scratch = torch.zeros(self.amax_blocks, dtype=torch.int64, device=dev)
        import os as _os
        self.attn_splits = int(_os.environ.get("DLLAMA_ATTN_SPLITS", "8"))
        # fused quantize-into-wire sync: validated bit-identical on hardware
        # (test_sync_quant_pack_matches_two_kernel_path); default on
        self.fused_sync = _os.environ.get("DLLAMA_FUSED_SYNC", "1") == "1"
        # adaptive K-split schedule: S=1 fused single-kernel attention below
        # pos 256 (no combine launch), S=8 to the threshold, S=16 beyond
        # (tools/attn_kv16_probe: S=16 wins from ~pos 512). The decode graph
        # is recaptured when pos crosses a boundary. DLLAMA_ADAPTIVE_SPLITS=0
        # (or an explicit DLLAMA_ATTN_SPLITS) pins S.
        self.adaptive_thresh = int(_os.environ.get("DLLAMA_ADAPTIVE_SPLITS", "512"))
        # measured-slower fusions kept for shape experiments (profiles r02):
        # the 16-wave fused FFN streams worse than the 4-wave GEMV + swiglu
        # pair, and per-wave gate recompute adds ~8 us to each MoE consumer
        self.fused_ffn = _os.environ.get("DLLAMA_FUSED_FFN", "0") == "1"
        self.fused_moe = _os.environ.get("DLLAMA_FUSED_MOE", "0") == "1"
        # deferred-quant dense decode (EPI_RESID_Q + PRO2); =0 reverts to
        # explicit norm_quant launches
        self.use_deferred = _os.environ.get("DLLAMA_DEFERRED", "1") == "1"
        # bf16 prefill: dense prompt chunks run hipBLASLt GEMMs (torch.mm)
        # over a bf16 copy of the weights. Measured SLOWER than the
        # hand-written int8-MFMA GEMM at the 32-token chunk shape (5355 vs
        # 5651 tok/s prefill) and costs a 2x-weight shadow — off by default,
        # kept as the library baseline for GEMM tuning A/Bs.
        self.prefill_bf16 = _os.environ.get("DLLAMA_PREFILL_BF16", "0") == "1"
        # K-split the deferred down-projections (wo/w2 underfill at 1 wg per
        # 32 rows when dim <= 4096); merged by add_ssq_q. A/B experiment.
        self.ksplit_resid = int(_os.environ.get("DLLAMA_KSPLIT_RESID", "0"))
        if ("DLLAMA_ATTN_SPLITS" in _os.environ
                and "DLLAMA_ADAPTIVE_SPLITS" not in _os.environ):
            self.adaptive_thresh = 0
        # split scratch sized for the current S (the adaptive schedule
        # reallocates on recapture: 8 short ctx, 16 past 512, 32 past 2k)
        self.attn_ml = torch.zeros(NB * c.n_heads0 * self.attn_splits * 2, device=dev)
        self.attn_o = torch.zeros(NB * c.n_heads0 * self.attn_splits * c.head_dim,
                                  device=dev)
        self.attn_counter = torch.zeros(NB * c.n_heads0, dtype=torch.int32, device=dev)
        if self.tp_path:
            # per-batch-size contiguous gather buffers: collectives need a
            # flat contiguous output (world, nb*...) — a [:, :nb] slice is not
            nbs = [n for n in (1, 2, 4, 8, 16, 32) if n <= NB]
            self.logits_gather = {n: torch.zeros(c.world, n, c.vocab0, device=dev)
                                  for n in nbs}
            self.logits_full = torch.zeros(NB, c.world * c.vocab0, device=dev)
            self.argmax_scratch_full = torch.zeros(
                -(-c.world * c.vocab0 // 4096), dtype=torch.int64, device=dev)
            if c.sync_type == Q80:
                row_bytes = c.dim + 2 * (c.dim // QB)
                self.sync_out = torch.zeros(NB * row_bytes, dtype=torch.uint8, device=dev)
                self.sync_in = {n: torch.zeros(c.world, n * row_bytes,
                                               dtype=torch.uint8, device=dev)
                                for n in nbs}
        if c.is_moe:
            S = NB * c.n_active_experts
            self.moe_idx = torch.zeros(S, dtype=torch.int32, device=dev)
            self.moe_wts = torch.zeros(NB, c.n_active_experts, device=dev)
            self.moe_router = torch.zeros(NB, c.n_experts, device=dev)
            self.moe_out13 = torch.zeros(S, 2 * c.ff_dim0, device=dev)
            self.moe_dq = QuantBuf(S, c.ff_dim0, dev)
            self.moe_y = torch.zeros(S, c.dim, device=dev)

    def _dequant_bf16(self, lin: Linear) -> torch.Tensor:
        """Q40 planes -> bf16 weight matrix on device (prefill GEMM shadow;
        same dequantized values the int8 GEMV streams, rounded to bf16)."""
        qs = lin.qs
        d, nb2 = qs.shape[-2], qs.shape[-1]
        n = nb2 * 2
        lo = (qs & 15).to(torch.float32) - 8.0
        hi = (qs >> 4).to(torch.float32) - 8.0
        lo = lo.view(*qs.shape[:-1], n // 32, 16)
        hi = hi.view(*qs.shape[:-1], n // 32, 16)
        w = torch.cat([lo, hi], dim=-1)  # elems j | j+16 per block
        sc = lin.scales.view(*qs.shape[:-1], n // 32, 1).float()
        return (w * sc).view(*qs.shape[:-2], d, n).to(torch.bfloat16)

    def _build_bf16_shadow(self):
        """bf16 copies of the dense matmul weights for hipBLASLt prefill
        (skipped for MoE/TP where the Q40 GEMM path stays)."""
        c = self.cfg
        if not self.prefill_bf16 or c.is_moe or self.tp_path:
            self.prefill_bf16 = False
            return
        self.wcls_bf16 = self._dequant_bf16(self.wcls)
        for lw in self.layers:
            lw["qkv_bf16"] = self._dequant_bf16(lw["qkv"])
            lw["wo_bf16"] = self._dequant_bf16(lw["wo"])
            lw["w13_bf16"] = self._dequant_bf16(lw["w13"])
            lw["w2_bf16"] = self._dequant_bf16(lw["w2"])

    def _finish_init(self):
        c, dev = self.cfg, self.device
        self._build_bf16_shadow()
        cache = R.rope_cache(c.seq_len, c.head_dim, c.rope_theta, c.rope_scaling)
        self.rope_cache = cache.reshape(c.seq_len, c.head_dim).contiguous().to(dev)
        # f16 KV cache by default: halves the attention HBM stream (measured
        # 1.45x at 1k ctx, 1.7x at 4k — tools/attn_kv16_probe; maxerr ~2e-5
        # vs f32). DLLAMA_KV_F32=1 reverts to the reference's f32 layout.
        import os as _os
        kv_dt = (torch.float32 if _os.environ.get("DLLAMA_KV_F32") == "1"
                 else torch.float16)
        self.k_cache = [torch.zeros(c.seq_len, c.kv_dim0, dtype=kv_dt, device=dev)
                        for _ in range(c.n_layers)]
        self.v_cache = [torch.zeros(c.seq_len, c.kv_dim0, dtype=kv_dt, device=dev)
                        for _ in range(c.n_layers)]
        self.rope_style = 1 if c.rope_type == ROPE_FALCON else 0

    # ------------------------------------------------------------ forward

    def _sync_partial(self, NB: int, slot: int):
        """TP>1: all-reduce self.partial[:NB] into x (+= sum of partials),
        accumulating the residual row's sum-of-squares into ssq[slot]."""
        c = self.cfg
        if c.sync_type == Q80:
            nb_dim = c.dim // QB
            row_bytes = c.dim + 2 * nb_dim
            out = self.sync_out[: NB * row_bytes]
            if getattr(self, "fused_sync", False):
                # quantize straight into the wire buffer, one pass
                # (default; DLLAMA_FUSED_SYNC=0 splits it)
                self.k.sync_quant_pack(self.partial[:NB], out)
            else:
                q = self.xq  # reuse dim-sized quant buffer
                self.k.q80_quantize(self.partial[:NB], q.q[:NB], q.s[:NB], q.bs[:NB])
                self.k.sync_pack(q.q[:NB], q.s[:NB], out)
            inb = self.sync_in[NB]
            self.comm.all_gather(inb, out)
            self.k.merge_add(self.x[:NB], inb.view(c.world, NB, row_bytes),
                             self.ssq[slot])
        else:
            self.comm.allreduce_(self.partial[:NB])
            self.k.add_ssq(self.x[:NB], self.partial[:NB], self.ssq[slot], NB)

    def _mm(self, lin: Linear, qb: QuantBuf, out, NB: int, amax=None):
        """Batched matmul dispatch: decode batches use the dot4 GEMV,
        prefill batches (>=8) the int8-MFMA GEMM."""
        if NB >= 8:
            self.k.q40_gemm(lin.qs, lin.scales, qb.q, qb.s, out, NB,
                            self.gemm_part)
        else:
            self.k.q40_gemv(lin.qs, lin.scales, qb.q, qb.s, qb.bs, out, NB, amax)

    def _proj_merge(self, lin: Linear, qb: QuantBuf, slot: int, NB: int):
        """Down-projection + residual fold: TP=1 decode fuses the add + ssq
        into the GEMV epilogue (+1us with cacheline-strided ssq slots — the
        earlier +6.5us was atomics serializing on one cacheline)."""
        k = self.k
        if not self.tp_path and NB < 8:
            k.q40_gemv_resid(lin.qs, lin.scales, qb.q, qb.s, qb.bs,
                             self.x, self.ssq[slot], NB)
            return
        self._mm(lin, qb, self.partial, NB)
        if not self.tp_path:
            k.add_ssq(self.x[:NB], self.partial[:NB], self.ssq[slot], NB)
        else:
            self._sync_partial(NB, slot)

    def forward_buffers(self, B: int):
        """Run one step over tokens[:B] at positions pos..pos+B-1, writing
        logits into logits0 (and the gather buffer under TP). Everything
        stays on device — this function is graph-capturable."""
        c, k = self.cfg, self.k
        NB = _pow2_batch(B)
        tp_def_ok = (not self.tp_path
                     or (c.sync_type == Q80 and c.dim % 256 == 0))
        if (B == 1 and c.dim % 32 == 0 and tp_def_ok
                and (not c.is_moe
                     or (c.dim % 256 == 0 and c.n_active_experts <= 16))
                and getattr(self, "use_deferred", True)
                and not getattr(self, "use_fused_norm", False)):
            return self._forward_dense_deferred()
        if B >= 8 and getattr(self, "prefill_bf16", False):
            return self._forward_prefill_bf16(B)
        x = self.x
        self.ssq.zero_()
        k.embed_gather(self.embedding, self.tokens, x, NB, self.ssq[0])

        kv_mul = c.n_heads0 // max(1, c.kv_dim0 // c.head_dim)
        fused_rope = self.rope_style == 0 and not c.is_qwen3
        # Fusing norm+quant into the GEMV prologue re-quantizes x per
        # workgroup: measured VALU-bound (2x slower on the big GEMVs).
        # Kept behind a flag for shapes where it might win; off by default.
        fused_norm = getattr(self, "use_fused_norm", False) and NB <= 4
        slot = 0

        def norm_gemv(lin, wn, slot, out, amax=None):
            """normed+quantized x -> matmul (fused prologue for decode)."""
            if fused_norm:
                k.q40_gemv_nq(lin.qs, lin.scales, x, wn, self.ssq[slot],
                              c.norm_eps, out, NB, amax)
            else:
                k.norm_quant(x[:NB], wn, self.ssq[slot], self.xq.q[:NB],
                             self.xq.s[:NB], self.xq.bs[:NB], NB, c.norm_eps)
                self._mm(lin, self.xq, out, NB, amax)

        for l, lw in enumerate(self.layers):
            # attention block
            if fused_rope and fused_norm:
                k.q40_gemv_nq_rope(lw["qkv"].qs, lw["qkv"].scales, x,
                                   lw["norm0"], self.ssq[slot], c.norm_eps,
                                   self.qkv_out, NB, self.rope_cache, self.pos,
                                   self.k_cache[l], self.v_cache[l],
                                   c.q_dim0, c.kv_dim0, c.head_dim)
            elif fused_rope and NB < 8:
                k.norm_quant(x[:NB], lw["norm0"], self.ssq[slot], self.xq.q[:NB],
                             self.xq.s[:NB], self.xq.bs[:NB], NB, c.norm_eps)
                k.q40_gemv_rope(lw["qkv"].qs, lw["qkv"].scales, self.xq.q,
                                self.xq.s, self.xq.bs, self.qkv_out, NB,
                                self.rope_cache, self.pos, self.k_cache[l],
                                self.v_cache[l], c.q_dim0, c.kv_dim0, c.head_dim)
            elif c.is_qwen3 and self.rope_style == 1:
                # per-head q/k rmsnorm + neox rope + KV write, one launch
                # (replaces 2x rmsnorm_rows_s + rope_kv = ~9 us/layer of
                # launch overhead in the captured graph)
                norm_gemv(lw["qkv"], lw["norm0"], slot, self.qkv_out)
                k.rope_kv_qknorm(self.qkv_out, self.qkv_ld, c.q_dim0,
                                 c.kv_dim0, self.rope_cache, self.pos,
                                 self.k_cache[l], self.v_cache[l], c.head_dim,
                                 lw["q_norm"], lw["k_norm"], c.norm_eps, B)
            else:
                norm_gemv(lw["qkv"], lw["norm0"], slot, self.qkv_out)
                if c.is_qwen3:
                    k.rmsnorm_rows_s(self.qkv_out, self.qkv_ld, 0,
                                     c.q_dim0 // c.head_dim, B, lw["q_norm"],
                                     c.head_dim, c.norm_eps)
                    k.rmsnorm_rows_s(self.qkv_out, self.qkv_ld, c.q_dim0,
                                     c.kv_dim0 // c.head_dim, B, lw["k_norm"],
                                     c.head_dim, c.norm_eps)
                k.rope_kv(self.qkv_out, self.qkv_ld, c.q_dim0, c.kv_dim0,
                          self.rope_cache, self.pos, self.k_cache[l],
                          self.v_cache[l], c.head_dim, self.rope_style, B)
            k.attn(self.qkv_out, self.qkv_ld, self.k_cache[l], self.v_cache[l],
                   self.zbuf[:B], self.pos, B, c.n_heads0, kv_mul, c.head_dim,
                   self.attn_splits, self.attn_ml, self.attn_o,
                   self.attn_counter, self.zq.q, self.zq.s, self.zq.bs)
            self._proj_merge(lw["wo"], self.zq, slot + 1, NB)
            slot += 1

            # ffn block
            if c.is_moe:
                # norm once: Q80 triple for the expert GEMVs + f32 for router
                k.norm_quant(x[:NB], lw["norm1"], self.ssq[slot], self.xq.q[:NB],
                             self.xq.s[:NB], self.xq.bs[:NB], NB, c.norm_eps,
                             self.t_norm[:NB])
                self._moe_ffn(B, NB, lw, slot + 1)
            else:
                gelu = c.hidden_act == HIDDEN_ACT_GELU
                if (NB == 1 and not fused_norm and c.ff_dim0 % 32 == 0
                        and getattr(self, "fused_ffn", False)):
                    # fused W1|W3 GEMV + SwiGLU + Q80 emit (one launch
                    # replacing gemv + swiglu_q80)
                    k.norm_quant(x[:NB], lw["norm1"], self.ssq[slot],
                                 self.xq.q[:NB], self.xq.s[:NB],
                                 self.xq.bs[:NB], NB, c.norm_eps)
                    k.q40_gemv_swiglu(lw["w13"].qs, lw["w13"].scales,
                                      self.xq.q, self.xq.s, self.xq.bs,
                                      self.dq.q, self.dq.s, self.dq.bs, gelu)
                else:
                    norm_gemv(lw["w13"], lw["norm1"], slot, self.ff_out)
                    k.swiglu_q80(self.ff_out, self.ff_out[:, c.ff_dim0:],
                                 2 * c.ff_dim0, c.ff_dim0, NB, self.dq.q[:NB],
                                 self.dq.s[:NB], self.dq.bs[:NB], gelu)
                self._proj_merge(lw["w2"], self.dq, slot + 1, NB)
            slot += 1

        if self.skip_logits and B > 1:
            return
        use_amax = (self.greedy_feedback and B == 1 and not self.tp_path)
        norm_gemv(self.wcls, self.final_norm, slot, self.logits0,
                  self.amax_scratch if use_amax else None)
        if self.tp_path:
            self.comm.all_gather(self.logits_gather[NB], self.logits0[:NB])
        if self.greedy_feedback and B == 1:
            # on-device greedy sampling feeding the next decode step (used by
            # the fully graph-captured bench loop; real serving samples on host)
            if use_amax:
                k.token_from_argmax(self.tokens, self.amax_scratch, self.amax_blocks)
            else:
                # TP: gathered slices are contiguous in global vocab order
                # (row-split wcls), so the flat gather buffer IS the full
                # logits row; two-kernel argmax, no ATen in the graph
                k.argmax_token(self.tokens, self.logits_gather[1].view(-1),
                               self.argmax_scratch_full)

    def _forward_prefill_bf16(self, B: int):
        """Dense prompt chunks as hipBLASLt bf16 GEMMs over the weight
        shadow: prefill is batch>=8 plain GEMM work, which belongs on the
        matrix cores — the hand-written Q40 int8 GEMM is VALU-bound at
        ~1.3 TB/s weight stream while the bf16 library GEMM is MFMA-bound
        (the reference's llamafile sgemm plays this exact role for batch>1,
        src/nn/llamafile/sgemm.cpp:819-986). rope/KV/attention/norm stay in
        the HIP kernels; elementwise glue is torch (prefill runs once per
        32-token chunk, not per token)."""
        import torch.nn.functional as F
        c, k = self.cfg, self.k
        NB = _pow2_batch(B)
        x = self.x
        eps = c.norm_eps
        self.ssq.zero_()
        k.embed_gather(self.embedding, self.tokens, x, NB, self.ssq[0])
        kv_mul = c.n_heads0 // max(1, c.kv_dim0 // c.head_dim)
        slot = 0
        for l, lw in enumerate(self.layers):
            k.norm_f32(x[:NB], lw["norm0"], self.ssq[slot], self.t_norm[:NB],
                       NB, eps)
            self.qkv_out[:NB].copy_(
                torch.mm(self.t_norm[:NB].bfloat16(), lw["qkv_bf16"].t()))
            if c.is_qwen3 and self.rope_style == 1:
                k.rope_kv_qknorm(self.qkv_out, self.qkv_ld, c.q_dim0,
                                 c.kv_dim0, self.rope_cache, self.pos,
                                 self.k_cache[l], self.v_cache[l], c.head_dim,
                                 lw["q_norm"], lw["k_norm"], eps, B)
            else:
                if c.is_qwen3:
                    k.rmsnorm_rows_s(self.qkv_out, self.qkv_ld, 0,
                                     c.q_dim0 // c.head_dim, B, lw["q_norm"],
                                     c.head_dim, eps)
                    k.rmsnorm_rows_s(self.qkv_out, self.qkv_ld, c.q_dim0,
                                     c.kv_dim0 // c.head_dim, B, lw["k_norm"],
                                     c.head_dim, eps)
                k.rope_kv(self.qkv_out, self.qkv_ld, c.q_dim0, c.kv_dim0,
                          self.rope_cache, self.pos, self.k_cache[l],
                          self.v_cache[l], c.head_dim, self.rope_style, B)
            k.attn(self.qkv_out, self.qkv_ld, self.k_cache[l], self.v_cache[l],
                   self.zbuf[:B], self.pos, B, c.n_heads0, kv_mul, c.head_dim,
                   self.attn_splits, self.attn_ml, self.attn_o,
                   self.attn_counter)
            partial = torch.mm(self.zbuf[:NB].bfloat16(),
                               lw["wo_bf16"].t()).float()
            k.add_ssq(x[:NB], partial, self.ssq[slot + 1], NB)
            slot += 1
            k.norm_f32(x[:NB], lw["norm1"], self.ssq[slot], self.t_norm[:NB],
                       NB, eps)
            ff = torch.mm(self.t_norm[:NB].bfloat16(), lw["w13_bf16"].t()).float()
            a, g = ff[:, :c.ff_dim0], ff[:, c.ff_dim0:]
            if c.hidden_act == HIDDEN_ACT_GELU:
                d = F.gelu(a, approximate="tanh") * g
            else:
                d = F.silu(a) * g
            partial = torch.mm(d.bfloat16(), lw["w2_bf16"].t()).float()
            k.add_ssq(x[:NB], partial, self.ssq[slot + 1], NB)
            slot += 1
        if self.skip_logits and B > 1:
            return
        k.norm_f32(x[:NB], self.final_norm, self.ssq[slot], self.t_norm[:NB],
                   NB, eps)
        self.logits0[:NB].copy_(
            torch.mm(self.t_norm[:NB].bfloat16(), self.wcls_bf16.t()))

    def _forward_dense_deferred(self):
        """B=1 dense decode with DEFERRED activation quantization: the
        down-projection GEMVs emit the next matmul's Q80 input in their own
        epilogue (x*w_norm quantized per wg-local block, scale excluding
        inv_rms), and every consumer GEMV applies inv = rsqrt(ssq/n + eps)
        as one multiply per output row. This removes both norm_quant
        launches per layer — the Q80 codes are scale-invariant, so numerics
        match the explicit-norm path to fp rounding (reference runs
        merge_add/inv_rms/rms_norm/cast as 4 ops per half-layer,
        llm.cpp:263-270)."""
        c, k = self.cfg, self.k
        x = self.x
        eps = c.norm_eps
        self.ssq.zero_()
        k.embed_gather(self.embedding, self.tokens, x, 1, self.ssq[0])
        kv_mul = c.n_heads0 // max(1, c.kv_dim0 // c.head_dim)
        fused_rope = self.rope_style == 0 and not c.is_qwen3
        # layer 0's norm0 quant comes from a deferred norm_quant (later
        # layers get it from the previous w2's EPI_RESID_Q epilogue)
        k.norm_quant(x[:1], self.layers[0]["norm0"], self.ssq[0],
                     self.xq.q[:1], self.xq.s[:1], self.xq.bs[:1], 1, eps,
                     deferred=True)
        slot = 0
        last = len(self.layers) - 1
        for l, lw in enumerate(self.layers):
            sin = self.ssq[slot]
            if fused_rope:
                k.q40_gemv_rope(lw["qkv"].qs, lw["qkv"].scales, self.xq.q,
                                self.xq.s, self.xq.bs, self.qkv_out, 1,
                                self.rope_cache, self.pos, self.k_cache[l],
                                self.v_cache[l], c.q_dim0, c.kv_dim0,
                                c.head_dim, ssq_in=sin, eps=eps)
            else:
                k.q40_gemv(lw["qkv"].qs, lw["qkv"].scales, self.xq.q,
                           self.xq.s, self.xq.bs, self.qkv_out, 1,
                           ssq_in=sin, eps=eps)
                if c.is_qwen3 and self.rope_style == 1:
                    k.rope_kv_qknorm(self.qkv_out, self.qkv_ld, c.q_dim0,
                                     c.kv_dim0, self.rope_cache, self.pos,
                                     self.k_cache[l], self.v_cache[l],
                                     c.head_dim, lw["q_norm"], lw["k_norm"],
                                     eps, 1)
                else:
                    if c.is_qwen3:
                        k.rmsnorm_rows_s(self.qkv_out, self.qkv_ld, 0,
                                         c.q_dim0 // c.head_dim, 1,
                                         lw["q_norm"], c.head_dim, eps)
                        k.rmsnorm_rows_s(self.qkv_out, self.qkv_ld, c.q_dim0,
                                         c.kv_dim0 // c.head_dim, 1,
                                         lw["k_norm"], c.head_dim, eps)
                    k.rope_kv(self.qkv_out, self.qkv_ld, c.q_dim0, c.kv_dim0,
                              self.rope_cache, self.pos, self.k_cache[l],
                              self.v_cache[l], c.head_dim, self.rope_style, 1)
            k.attn(self.qkv_out, self.qkv_ld, self.k_cache[l], self.v_cache[l],
                   self.zbuf[:1], self.pos, 1, c.n_heads0, kv_mul, c.head_dim,
                   self.attn_splits, self.attn_ml, self.attn_o,
                   self.attn_counter, self.zq.q, self.zq.s, self.zq.bs)
            # wo: residual fold + deferred Q80 emit of x*norm1 for the FFN
            # (TP: the epilogue packs the Q80 wire instead; merge-add after
            # the all-gather folds the residual and emits the deferred quant
            # — reference SYNC_NODE_SLICES + OP_MERGE_ADD, one fewer launch
            # on each side of the collective)
            if self.tp_path:
                self._tp_proj_deferred(lw["wo"], self.zq, self.ssq[slot + 1],
                                       lw["norm1"])
            elif self.ksplit_resid:
                k.q40_gemv_ksplit(lw["wo"].qs, lw["wo"].scales, self.zq.q,
                                  self.zq.s, self.zq.bs, self.gemm_part,
                                  self.ksplit_resid)
                k.add_ssq_q(x[:1], self.gemm_part, self.ssq[slot + 1],
                            lw["norm1"], self.xq.q[:1], self.xq.s[:1],
                            self.xq.bs[:1], self.ksplit_resid)
            else:
                k.q40_gemv_resid_q(lw["wo"].qs, lw["wo"].scales, self.zq.q,
                                   self.zq.s, self.zq.bs, x,
                                   self.ssq[slot + 1], lw["norm1"],
                                   self.xq.q, self.xq.s, self.xq.bs)
            slot += 1
            wn = self.final_norm if l == last else self.layers[l + 1]["norm0"]
            if c.is_moe:
                # MoE deferred FFN: router reads x*norm1*inv directly (no
                # t_norm buffer), grouped w13 consumes the deferred xq, and
                # the scale-merge epilogue emits the NEXT layer's deferred
                # quant — both per-layer norm_quant launches gone here too
                ka = c.n_active_experts
                k.router_gemv_norm(lw["gate"], x, lw["norm1"],
                                   self.ssq[slot], eps, self.moe_router, 1)
                k.moe_gate(self.moe_router[:1], self.moe_idx, self.moe_wts,
                           1, ka)
                k.q40_gemv_grouped(lw["w13"].qs, lw["w13"].scales, self.xq.q,
                                   self.xq.s, self.xq.bs, self.moe_idx[:ka],
                                   self.moe_out13, ka,
                                   ssq_in=self.ssq[slot], eps=eps)
                k.swiglu_q80(self.moe_out13, self.moe_out13[:, c.ff_dim0:],
                             2 * c.ff_dim0, c.ff_dim0, ka, self.moe_dq.q[:ka],
                             self.moe_dq.s[:ka], self.moe_dq.bs[:ka])
                k.q40_gemv_grouped(lw["w2"].qs, lw["w2"].scales, self.moe_dq.q,
                                   self.moe_dq.s, self.moe_dq.bs,
                                   self.moe_idx[:ka], self.moe_y, 1)
                if self.tp_path:
                    # expert-weighted sum packed straight into the Q80 wire;
                    # merge-add after the gather folds residual + emits the
                    # next deferred quant
                    rb = c.dim + 2 * (c.dim // QB)
                    k.scale_merge_pack(self.moe_y, self.moe_wts,
                                       self.sync_out[:rb], 1, ka, c.dim)
                    inb = self.sync_in[1]
                    self.comm.all_gather(inb, self.sync_out[:rb])
                    k.merge_add_q(x[:1], inb.view(c.world, 1, rb),
                                  self.ssq[slot + 1], wn, self.xq.q,
                                  self.xq.s, self.xq.bs)
                else:
                    k.scale_merge_add_q(x[:1], self.moe_y, self.moe_wts,
                                        self.ssq[slot + 1], wn, self.xq.q,
                                        self.xq.s, self.xq.bs, 1, ka)
            else:
                k.q40_gemv(lw["w13"].qs, lw["w13"].scales, self.xq.q,
                           self.xq.s, self.xq.bs, self.ff_out, 1,
                           ssq_in=self.ssq[slot], eps=eps)
                k.swiglu_q80(self.ff_out, self.ff_out[:, c.ff_dim0:],
                             2 * c.ff_dim0, c.ff_dim0, 1, self.dq.q[:1],
                             self.dq.s[:1], self.dq.bs[:1],
                             c.hidden_act == HIDDEN_ACT_GELU)
                # w2: residual fold + deferred emit for the NEXT layer's
                # norm0 (final_norm for logits after the last layer)
                if self.tp_path:
                    self._tp_proj_deferred(lw["w2"], self.dq,
                                           self.ssq[slot + 1], wn)
                elif self.ksplit_resid:
                    k.q40_gemv_ksplit(lw["w2"].qs, lw["w2"].scales, self.dq.q,
                                      self.dq.s, self.dq.bs, self.gemm_part,
                                      self.ksplit_resid)
                    k.add_ssq_q(x[:1], self.gemm_part, self.ssq[slot + 1],
                                wn, self.xq.q[:1], self.xq.s[:1],
                                self.xq.bs[:1], self.ksplit_resid)
                else:
                    k.q40_gemv_resid_q(lw["w2"].qs, lw["w2"].scales,
                                       self.dq.q, self.dq.s, self.dq.bs, x,
                                       self.ssq[slot + 1], wn, self.xq.q,
                                       self.xq.s, self.xq.bs)
            slot += 1
        use_amax = self.greedy_feedback and not self.tp_path
        k.q40_gemv(self.wcls.qs, self.wcls.scales, self.xq.q, self.xq.s,
                   self.xq.bs, self.logits0, 1,
                   self.amax_scratch if use_amax else None,
                   ssq_in=self.ssq[slot], eps=eps)
        if self.tp_path:
            self.comm.all_gather(self.logits_gather[1], self.logits0[:1])
        if self.greedy_feedback:
            if use_amax:
                k.token_from_argmax(self.tokens, self.amax_scratch,
                                    self.amax_blocks)
            else:
                k.argmax_token(self.tokens, self.logits_gather[1].view(-1),
                               self.argmax_scratch_full)

    def _tp_proj_deferred(self, lin: Linear, qb: QuantBuf, ssq_slot,
                          wnorm: torch.Tensor):
        """TP down-projection, deferred form: the GEMV epilogue emits the
        Q80 wire directly, the all-gathered slices merge into x with the
        next matmul's deferred quant emitted in the same kernel."""
        c, k = self.cfg, self.k
        row_bytes = c.dim + 2 * (c.dim // QB)
        k.q40_gemv_pack(lin.qs, lin.scales, qb.q, qb.s, qb.bs,
                        self.sync_out[:row_bytes])
        inb = self.sync_in[1]
        self.comm.all_gather(inb, self.sync_out[:row_bytes])
        k.merge_add_q(self.x[:1], inb.view(c.world, 1, row_bytes), ssq_slot,
                      wnorm, self.xq.q, self.xq.s, self.xq.bs)

    def _moe_ffn(self, B: int, NB: int, lw: dict, slot: int):
        """Router + grouped expert GEMVs (reference llm.cpp:450-487);
        t_norm holds the f32 normed activations (router input), xq the same
        values Q80-quantized (expert GEMV input = reference repeat_z)."""
        c, k = self.cfg, self.k
        ka = c.n_active_experts
        S = NB * ka
        k.router_gemv(lw["gate"], self.t_norm, self.moe_router, NB)
        if (c.ff_dim0 % 32 == 0 and ka <= 16
                and getattr(self, "fused_moe", False)):
            # gate-fused decode path: every consumer recomputes the
            # deterministic top-k from the router logits in-kernel, so the
            # FFN is 3 launches (w13+swiglu, w2, scale-merge) instead of 6
            router = self.moe_router[:NB]
            k.q40_gemv_grouped_swiglu(lw["w13"].qs, lw["w13"].scales,
                                      self.xq.q, self.xq.s, self.xq.bs,
                                      self.moe_dq.q, self.moe_dq.s,
                                      self.moe_dq.bs, S, router, ka)
            k.q40_gemv_grouped(lw["w2"].qs, lw["w2"].scales, self.moe_dq.q,
                               self.moe_dq.s, self.moe_dq.bs,
                               self.moe_idx[:S], self.moe_y, 1,
                               router=router, topk=ka, n_slots=S)
            if not self.tp_path:
                k.scale_merge_add(self.x[:NB], self.moe_y, router,
                                  self.ssq[slot], NB, ka, gate=True)
            else:
                k.scale_merge(self.partial[:NB], self.moe_y, router, NB, ka,
                              gate=True)
                self._sync_partial(NB, slot)
            return
        k.moe_gate(self.moe_router[:NB], self.moe_idx, self.moe_wts, NB, ka)
        k.q40_gemv_grouped(lw["w13"].qs, lw["w13"].scales, self.xq.q, self.xq.s,
                           self.xq.bs, self.moe_idx[:S], self.moe_out13, ka)
        k.swiglu_q80(self.moe_out13, self.moe_out13[:, c.ff_dim0:],
                     2 * c.ff_dim0, c.ff_dim0, S, self.moe_dq.q[:S],
                     self.moe_dq.s[:S], self.moe_dq.bs[:S])
        k.q40_gemv_grouped(lw["w2"].qs, lw["w2"].scales, self.moe_dq.q,
                           self.moe_dq.s, self.moe_dq.bs, self.moe_idx[:S],
                           self.moe_y, 1)
        if not self.tp_path:
            k.scale_merge_add(self.x[:NB], self.moe_y, self.moe_wts,
                              self.ssq[slot], NB, ka)
        else:
            k.scale_merge(self.partial[:NB], self.moe_y, self.moe_wts, NB, ka)
            self._sync_partial(NB, slot)

    # ------------------------------------------------------------ engine API

    def forward(self, tokens: torch.Tensor, positions: torch.Tensor) -> torch.Tensor:
        """Engine-compatible forward. positions must be a contiguous range.

        When a decode graph is captured and B==1, this is a single hipGraph
        replay (the graph advances the device position itself)."""
        B = tokens.shape[0]
        assert B <= self.n_batches
        p0 = int(positions[0])
        if p0 + B > self.cfg.seq_len:
            raise ValueError(
                f"position {p0}+{B} exceeds seq_len {self.cfg.seq_len} "
                "(rebuild with a larger --max-seq-len / seq_len)")
        self.tokens[:B].copy_(tokens.to(self.device), non_blocking=True)
        if B == 1 and self._graph is not None:
            sp = self._pick_splits(p0)
            if sp != self.attn_splits:
                self._set_attn_splits(sp)  # recapture at the new K-split count
            if p0 != self._graph_pos:
                self.pos.fill_(p0)
            self._graph.replay()
            self._graph_pos = p0 + 1  # the graph's pos_inc advanced it
        elif B == self.n_batches and not self._pf_failed:
            # full prefill chunks replay a captured graph (eager chunk cost
            # is host-launch-bound: ~400 python kernel launches)
            g = self._pf_graphs.get(self.skip_logits)
            if g is None:
                g = self._capture_prefill_graph(self.skip_logits)
            self.pos.fill_(p0)
            self._graph_pos = None
            if g is not None:
                g.replay()
            else:
                self.forward_buffers(B)
        else:
            self.pos.fill_(p0)
            self._graph_pos = None
            self.forward_buffers(B)
        c = self.cfg
        if self.tp_path:
            NBp = _pow2_batch(B)
            self.k.logits_concat(self.logits_full, self.logits_gather[NBp], B)
            return self.logits_full[:B]
        return self.logits0[:B]

    # ------------------------------------------------------------ graphs

    def _pick_splits(self, pos: int) -> int:
        # S=8 short context, S=16 past the threshold (tools/attn_kv16_probe).
        # Two measured-slower variants exist behind env flags and are never
        # auto-picked: the S=1 fused single kernel (13.5 us vs the ~11 us
        # pair) and the GQA-grouped split (DLLAMA_GQA_ATTN).
        if not self.adaptive_thresh:
            return self.attn_splits
        if pos < self.adaptive_thresh:
            return 8
        return 16 if pos < 2048 else 32  # S=32 at 4k ctx: +12.5% same-box

    def _set_attn_splits(self, s: int):
        """Switch the flash-decode K-split count mid-stream (long-context
        adaptivity): resize the split scratch and recapture the decode
        graph. Decode state (KV caches, device pos) is untouched; the
        capture warmup's KV write at the current pos is overwritten by the
        next real step."""
        c = self.cfg
        self.attn_splits = s
        NB = self.n_batches
        dev = self.device
        self.attn_ml = torch.zeros(NB * c.n_heads0 * s * 2, device=dev)
        self.attn_o = torch.zeros(NB * c.n_heads0 * s * c.head_dim, device=dev)
        if self._graph is not None:
            pos_saved = int(self.pos.item())
            self.capture_decode_graph()
            self.pos.fill_(pos_saved)
            self._graph_pos = pos_saved
        self._pf_graphs.clear()  # prefill graphs also reference the scratch

    def _capture_prefill_graph(self, skip_logits: bool):
        try:
            torch.cuda.synchronize(self.device)
            saved = self.skip_logits
            self.skip_logits = skip_logits
            s = torch.cuda.Stream(self.device)
            s.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(s):
                self.forward_buffers(self.n_batches)
            torch.cuda.current_stream(self.device).wait_stream(s)
            torch.cuda.synchronize(self.device)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self.forward_buffers(self.n_batches)
            self.skip_logits = saved
            self._pf_graphs[skip_logits] = g
            return g
        except Exception:  # noqa: BLE001
            self._pf_failed = True
            self.skip_logits = skip_logits
            return None

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
