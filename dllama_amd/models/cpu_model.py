"""CPU (PyTorch fp32) transformer — the numerics oracle and the no-GPU path
(BASELINE config 1: Llama-3.2-1B `dllama inference` on CPU).

Computation order parity with the reference per-layer step stream
(llm.cpp:263-557, summarized in SURVEY.md §3.1):
  merge_add -> rms_norm -> cast(q80) -> q,k,v matmul -> [qwen3 q/k norm] ->
  rope -> kv append -> attention -> cast -> wo matmul -> SYNC ->
  merge_add -> rms_norm -> [ffn | moe] -> SYNC
final: merge_add -> final norm -> cast -> logits matmul -> gather.

The Q80 activation casts are modeled as quantize->dequantize round-trips so
the CPU path sees the same quantization error profile as the HIP kernels.
Q80 sync is modeled as a per-rank partial round-trip before the all-reduce
(numerically identical to all-gather + dequant + merge-add).
"""

from __future__ import annotations

import numpy as np
import torch

from ..model_file import HIDDEN_ACT_GELU, ModelFile, ROPE_FALCON
from ..ops import reference as R
from ..parallel.comm import Comm, SingleComm
from ..quants import Q40, Q80
from .config import ModelConfig


class Q40W:
    """A Q40 weight kept in plane layout on CPU (uint8 nibbles + f16
    scales); the native extension op streams it without a f32 copy."""

    def __init__(self, qs: torch.Tensor, scales: torch.Tensor):
        self.qs = qs
        self.scales = scales
        self.d = qs.shape[-2]

    def __getitem__(self, e: int) -> "Q40W":  # stacked MoE experts
        return Q40W(self.qs[e], self.scales[e])


class CpuTransformer:
    def __init__(self, m: ModelFile, config: ModelConfig, comm: Comm | None = None,
                 activation_quant: bool = True, weight_dtype: torch.dtype = torch.float32):
        self.cfg = config
        self.comm = comm or SingleComm()
        self.activation_quant = activation_quant  # False = pure f32 (debugging)
        self.skip_logits = False  # accepted for engine compat; CPU always computes
        # f16 weights halve resident memory (useful for big models on small
        # hosts); measured speed-neutral on torch CPU (interleaved A/B, 1B
        # decode: 0.92x). f32 stays the default: it is the numerics oracle
        # and the reference-parity path.
        self.weight_dtype = weight_dtype
        self.q40_native = weight_dtype == "q40"
        if self.q40_native:
            # native Q40 streaming: quantized-weight RAM instead of the 8x
            # f32 copy (reference CPU path, nn-cpu-ops.cpp:231-449)
            if m.header.weight_type != Q40:
                raise ValueError("--cpu-dtype q40 needs a Q40 .m file")
            from ..ops import hip_ops
            self._k = hip_ops()  # the in-tree extension (CPU entry point)
        c = self.cfg
        r, w = c.rank, c.world

        def t(name, layer=-1, expert=-1):
            if self.q40_native:
                qs, sc = m.slice_q40_planes(name, layer, r, w, expert)
                return Q40W(torch.from_numpy(qs), torch.from_numpy(sc))
            return torch.from_numpy(
                np.array(m.slice_f32(name, layer, r, w, expert))).to(weight_dtype)

        self.embedding = torch.from_numpy(np.array(m.f32("embedding")))
        self.final_norm = torch.from_numpy(np.array(m.f32("final_norm")))
        self.wcls = t("final_matmul_logits")
        self.layers = []
        for l in range(c.n_layers):
            lw = {
                "q": t("block_matmul_q", l), "k": t("block_matmul_k", l),
                "v": t("block_matmul_v", l), "wo": t("block_matmul_wo", l),
                "norm0": torch.from_numpy(np.array(m.f32("block_norm_0", l))),
                "norm1": torch.from_numpy(np.array(m.f32("block_norm_1", l))),
            }
            if c.is_moe:
                lw["gate"] = torch.from_numpy(np.array(m.f32("block_moe_gate", l)))
                def stack(name):
                    parts = [t(name, l, e) for e in range(c.n_experts)]
                    if self.q40_native:
                        return Q40W(torch.stack([p.qs for p in parts]),
                                    torch.stack([p.scales for p in parts]))
                    return torch.stack(parts)
                lw["w1"] = stack("block_matmul_w1")
                lw["w2"] = stack("block_matmul_w2")
                lw["w3"] = stack("block_matmul_w3")
            else:
                lw["w1"] = t("block_matmul_w1", l)
                lw["w2"] = t("block_matmul_w2", l)
                lw["w3"] = t("block_matmul_w3", l)
            if c.is_qwen3:
                lw["q_norm"] = torch.from_numpy(np.array(m.f32("block_norm_q", l)))
                lw["k_norm"] = torch.from_numpy(np.array(m.f32("block_norm_k", l)))
            self.layers.append(lw)

        self.rope = R.rope_cache(c.seq_len, c.head_dim, c.rope_theta, c.rope_scaling)
        self.k_cache = torch.zeros(c.n_layers, c.seq_len, c.kv_dim0)
        self.v_cache = torch.zeros(c.n_layers, c.seq_len, c.kv_dim0)

    # ------------------------------------------------------------------

    def _sync(self, partial: torch.Tensor) -> torch.Tensor:
        if self.cfg.sync_type == Q80 and self.cfg.world > 1 and self.activation_quant:
            partial = R.q80_roundtrip(partial)
        return self.comm.allreduce_(partial)

    def _matmul(self, x: torch.Tensor, w) -> torch.Tensor:
        if isinstance(w, Q40W):
            if self.activation_quant:
                x = R.q80_roundtrip(x)
            y = torch.zeros(x.shape[0], w.d)
            self._k.q40_matmul_cpu(w.qs, w.scales, x.contiguous().float(), y)
            return y
        if self.weight_dtype != torch.float32:
            # fast serving path: Q80-roundtripped activations (same wire
            # semantics), f16 weight stream, f32 output
            if self.activation_quant:
                x = R.q80_roundtrip(x)
            return (x.to(self.weight_dtype) @ w.t()).float()
        return R.q40_matmul(x, w, quantize_x=self.activation_quant)

    def _rope(self, x: torch.Tensor, positions: torch.Tensor) -> torch.Tensor:
        if self.cfg.rope_type == ROPE_FALCON:
            return R.rope_falcon(x, self.rope, positions, self.cfg.head_dim)
        return R.rope_llama(x, self.rope, positions, self.cfg.head_dim)

    def forward(self, tokens: torch.Tensor, positions: torch.Tensor) -> torch.Tensor:
        """tokens, positions: int64 [B] -> logits f32 [B, vocab] (all ranks)."""
        c = self.cfg
        B = tokens.shape[0]
        if int(positions[-1]) >= c.seq_len:
            raise ValueError(
                f"position {int(positions[-1])} exceeds seq_len {c.seq_len}")
        x = self.embedding[tokens.long()].clone()  # [B, dim], replicated

        for l, lw in enumerate(self.layers):
            t0 = R.rms_norm(x, lw["norm0"], c.norm_eps)
            q = self._matmul(t0, lw["q"])
            k = self._matmul(t0, lw["k"])
            v = self._matmul(t0, lw["v"])
            if c.is_qwen3:
                q = R.rms_norm(q.reshape(B, -1, c.head_dim), lw["q_norm"],
                               c.norm_eps).reshape(B, -1)
                k = R.rms_norm(k.reshape(B, -1, c.head_dim), lw["k_norm"],
                               c.norm_eps).reshape(B, -1)
            q = self._rope(q, positions)
            k = self._rope(k, positions)
            self.k_cache[l, positions.long()] = k
            self.v_cache[l, positions.long()] = v
            z = R.attention(q, self.k_cache[l], self.v_cache[l], positions,
                            c.n_heads0, c.head_dim)
            partial = self._matmul(z, lw["wo"])
            x = x + self._sync(partial)

            t1 = R.rms_norm(x, lw["norm1"], c.norm_eps)
            if c.is_moe:
                partial = self._moe_ffn(t1, lw)
            else:
                a = self._matmul(t1, lw["w1"])
                g = self._matmul(t1, lw["w3"])
                if c.hidden_act == HIDDEN_ACT_GELU:
                    d = R.gelu(a) * g
                else:
                    d = R.swiglu(a, g)
                partial = self._matmul(d, lw["w2"])
            x = x + self._sync(partial)

        t = R.rms_norm(x, self.final_norm, c.norm_eps)
        logits0 = self._matmul(t, self.wcls)  # [B, vocab0]
        if c.world == 1:
            return logits0
        out = torch.empty(c.world, B, c.vocab0)
        self.comm.all_gather(out, logits0)
        return out.permute(1, 0, 2).reshape(B, c.vocab_size)

    def _moe_ffn(self, t1: torch.Tensor, lw: dict) -> torch.Tensor:
        """TP-sharded MoE FFN: every rank holds a slice of ALL experts,
        gate computed redundantly per rank (reference llm.cpp:450-487,
        SURVEY.md §2.2 EP row)."""
        c = self.cfg
        B = t1.shape[0]
        router = t1 @ lw["gate"].t()
        idx, wts = R.moe_gate(router, c.n_active_experts)  # [B,k]
        partial = torch.zeros(B, c.dim)
        for b in range(B):
            for s in range(c.n_active_experts):
                e = int(idx[b, s])
                a = self._matmul(t1[b:b + 1], lw["w1"][e])[0]
                g = self._matmul(t1[b:b + 1], lw["w3"][e])[0]
                d = R.swiglu(a, g)
                partial[b] += wts[b, s] * self._matmul(d.reshape(1, -1),
                                                       lw["w2"][e])[0]
        return partial
