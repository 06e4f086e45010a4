"""Model + TP-shard configuration derived from a .m header.

Sharding math parity with the reference slicers (nn-core.cpp:211-285):
  - q/k/v/w1/w3/wcls row-split: rank holds d/world output rows
  - wo/w2 col-split: rank holds n/world input cols, produces full-dim partial
  - heads split n_heads/world, kv cache split kv_dim/world
  - constraint: world <= n_kv_heads and world a power of 2 (app.cpp:236-238)
"""

from __future__ import annotations

from dataclasses import dataclass

from ..model_file import (ARCH_QWEN3, ARCH_QWEN3_MOE, LlmHeader, ROPE_LLAMA3_1)


@dataclass
class ModelConfig:
    arch_type: int
    dim: int
    hidden_dim: int
    ff_dim: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    head_dim: int
    n_experts: int
    n_active_experts: int
    vocab_size: int
    seq_len: int
    rope_type: int
    rope_theta: float
    rope_scaling: dict | None
    norm_eps: float
    hidden_act: int
    sync_type: int

    world: int = 1
    rank: int = 0

    @classmethod
    def from_header(cls, h: LlmHeader, world: int = 1, rank: int = 0) -> "ModelConfig":
        if world > 1:
            assert world & (world - 1) == 0, "world size must be a power of two"
            assert world <= h.n_kv_heads, \
                f"world {world} > n_kv_heads {h.n_kv_heads} (reference app.cpp:236)"
        scaling = None
        if h.rope_type == ROPE_LLAMA3_1 and h.rope_scaling_factor != 1.0:
            scaling = dict(factor=h.rope_scaling_factor,
                           low_freq_factor=h.rope_scaling_low_freq_factor,
                           high_freq_factor=h.rope_scaling_high_freq_factor,
                           orig_max_seq_len=h.rope_scaling_orig_max_seq_len)
        return cls(arch_type=h.arch_type, dim=h.dim, hidden_dim=h.hidden_dim,
                   ff_dim=h.ff_dim, n_layers=h.n_layers, n_heads=h.n_heads,
                   n_kv_heads=h.n_kv_heads, head_dim=h.head_dim,
                   n_experts=h.n_experts, n_active_experts=h.n_active_experts,
                   vocab_size=h.vocab_size, seq_len=h.seq_len,
                   rope_type=h.rope_type, rope_theta=h.rope_theta,
                   rope_scaling=scaling, norm_eps=h.norm_epsilon,
                   hidden_act=h.hidden_act, sync_type=h.sync_type,
                   world=world, rank=rank)

    # per-rank (sliced) dims
    @property
    def q_dim(self) -> int:
        return self.n_heads * self.head_dim

    @property
    def kv_dim(self) -> int:
        return self.n_kv_heads * self.head_dim

    @property
    def q_dim0(self) -> int:
        return self.q_dim // self.world

    @property
    def kv_dim0(self) -> int:
        return self.kv_dim // self.world

    @property
    def n_heads0(self) -> int:
        return self.n_heads // self.world

    @property
    def n_kv_heads0(self) -> int:
        return max(1, self.n_kv_heads // self.world)

    @property
    def ff_dim0(self) -> int:
        return self.ff_dim // self.world

    @property
    def vocab0(self) -> int:
        return self.vocab_size // self.world

    @property
    def is_qwen3(self) -> bool:
        return self.arch_type in (ARCH_QWEN3, ARCH_QWEN3_MOE)

    @property
    def is_moe(self) -> bool:
        return self.arch_type == ARCH_QWEN3_MOE and self.n_experts > 0
