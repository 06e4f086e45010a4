"""`.m` model file format reader/writer.

Format parity with the reference:
  - header: magic 0xA00ABCD (i32), headerSize (i32, includes the 8 bytes of
    magic+size), then (key, value) i32 pairs — reference src/llm.cpp:36-116,
    converter/writer.py:109-148. Weight data starts at offset headerSize.
  - weight walk order: embedding; per layer q,k,v,wo,[moe gate,
    experts x (w1,w2,w3) | w1,w2,w3], [qwen3 q/k norm], norm0, norm1;
    final norm; wcls — reference src/llm.cpp:614-669.

Matmul weights are stored row-major (d_out, n_in) in the file, in the
weight float type (Q40 for the shipped models); norms/embeddings in f32.
"""

from __future__ import annotations

import io
import struct
from dataclasses import dataclass, field

import numpy as np

from . import quants
from .quants import F32, Q40, Q80, tensor_bytes

MODEL_MAGIC = 0xA00ABCD

# header keys (reference src/llm.hpp:8-31)
K_VERSION = 0
K_ARCH_TYPE = 1
K_DIM = 2
K_HIDDEN_DIM = 3
K_N_LAYERS = 4
K_N_HEADS = 5
K_N_KV_HEADS = 6
K_N_EXPERTS = 7
K_N_ACTIVE_EXPERTS = 8
K_VOCAB_SIZE = 9
K_SEQ_LEN = 10
K_HIDDEN_ACT = 11
K_ROPE_THETA = 12
K_WEIGHT_FLOAT_TYPE = 13
K_ROPE_SCALING_FACTOR = 14
K_ROPE_SCALING_LOW_FREQ_FACTOR = 15
K_ROPE_SCALING_HIGH_FREQ_FACTORY = 16
K_ROPE_SCALING_ORIG_MAX_SEQ_LEN = 17
K_ROPE_TYPE = 18
K_HEAD_DIM = 19
K_NORM_EPSILON = 20
K_MOE_HIDDEN_DIM = 21

# arch types (reference src/llm.hpp:38-42)
ARCH_LLAMA = 0xABCD00
ARCH_QWEN3 = 0xABCD01
ARCH_QWEN3_MOE = 0xABCD02

# rope types (reference src/nn/nn-core.hpp)
ROPE_LLAMA = 0
ROPE_FALCON = 1
ROPE_LLAMA3_1 = 2

HIDDEN_ACT_GELU = 0
HIDDEN_ACT_SILU = 1


@dataclass
class LlmHeader:
    """Parsed .m header (reference src/llm.hpp:44-74)."""

    arch_type: int = ARCH_LLAMA
    version: int = 0
    dim: int = 0
    hidden_dim: int = 0
    n_layers: int = 0
    n_heads: int = 0
    n_kv_heads: int = 0
    n_experts: int = 0
    n_active_experts: int = 0
    vocab_size: int = 0
    seq_len: int = 0
    orig_seq_len: int = 0
    hidden_act: int = HIDDEN_ACT_SILU
    rope_theta: float = 10000.0
    weight_type: int = Q40
    rope_scaling_factor: float = 1.0
    rope_scaling_low_freq_factor: float = 0.0
    rope_scaling_high_freq_factor: float = 0.0
    rope_scaling_orig_max_seq_len: int = 0
    rope_type: int = ROPE_LLAMA
    head_dim: int = 0
    norm_epsilon: float = 1e-5
    moe_hidden_dim: int = 0
    sync_type: int = Q80
    header_size: int = 0
    file_size: int = 0

    def finalize(self, max_seq_len: int = 0) -> "LlmHeader":
        """Derived fields + seq-len clamp (reference src/llm.cpp:102-115)."""
        self.orig_seq_len = self.seq_len
        if max_seq_len and self.seq_len > max_seq_len:
            self.seq_len = max_seq_len
        if self.head_dim == 0:
            self.head_dim = self.dim // self.n_heads
        if self.arch_type in (ARCH_QWEN3, ARCH_QWEN3_MOE):
            self.rope_type = ROPE_FALCON
        return self

    @property
    def q_dim(self) -> int:
        return self.head_dim * self.n_heads

    @property
    def kv_dim(self) -> int:
        return self.head_dim * self.n_kv_heads

    @property
    def ff_dim(self) -> int:
        """Per-expert FFN width (moe_hidden_dim for MoE; llm.cpp:154-157)."""
        return self.moe_hidden_dim if self.arch_type == ARCH_QWEN3_MOE else self.hidden_dim

    @property
    def is_qwen3(self) -> bool:
        return self.arch_type in (ARCH_QWEN3, ARCH_QWEN3_MOE)


def _norm_eps_to_int(eps: float) -> int:
    if abs(eps - 1e-5) < 1e-12:
        return 5
    if abs(eps - 1e-6) < 1e-13:
        return 6
    raise ValueError(f"unsupported norm epsilon {eps}")


def _int_to_norm_eps(v: int) -> float:
    if v == 5:
        return 1e-5
    if v == 6:
        return 1e-6
    raise ValueError(f"unsupported norm epsilon code {v}")


def read_header(path: str, max_seq_len: int = 0, sync_type: int = Q80) -> LlmHeader:
    """Parse a .m header (reference src/llm.cpp:36-116)."""
    h = LlmHeader(sync_type=sync_type)
    with open(path, "rb") as f:
        magic, header_size = struct.unpack("<ii", f.read(8))
        if magic != MODEL_MAGIC:
            raise ValueError(f"unsupported magic 0x{magic:X} in {path}")
        h.header_size = header_size
        data = f.read(header_size - 8)
        f.seek(0, io.SEEK_END)
        h.file_size = f.tell()
    kv = struct.unpack(f"<{len(data) // 4}i", data)
    for key, value in zip(kv[0::2], kv[1::2]):
        if key == K_VERSION:
            h.version = value
        elif key == K_ARCH_TYPE:
            h.arch_type = value
        elif key == K_DIM:
            h.dim = value
        elif key == K_HIDDEN_DIM:
            h.hidden_dim = value
        elif key == K_N_LAYERS:
            h.n_layers = value
        elif key == K_N_HEADS:
            h.n_heads = value
        elif key == K_N_KV_HEADS:
            h.n_kv_heads = value
        elif key == K_N_EXPERTS:
            h.n_experts = value
        elif key == K_N_ACTIVE_EXPERTS:
            h.n_active_experts = value
        elif key == K_VOCAB_SIZE:
            h.vocab_size = value
        elif key == K_SEQ_LEN:
            h.seq_len = value
        elif key == K_HIDDEN_ACT:
            h.hidden_act = value
        elif key == K_ROPE_THETA:
            h.rope_theta = float(value)
        elif key == K_WEIGHT_FLOAT_TYPE:
            h.weight_type = value
        elif key == K_ROPE_SCALING_FACTOR:
            h.rope_scaling_factor = float(value)
        elif key == K_ROPE_SCALING_LOW_FREQ_FACTOR:
            h.rope_scaling_low_freq_factor = float(value)
        elif key == K_ROPE_SCALING_HIGH_FREQ_FACTORY:
            h.rope_scaling_high_freq_factor = float(value)
        elif key == K_ROPE_SCALING_ORIG_MAX_SEQ_LEN:
            h.rope_scaling_orig_max_seq_len = value
        elif key == K_ROPE_TYPE:
            h.rope_type = value
        elif key == K_HEAD_DIM:
            h.head_dim = value
        elif key == K_NORM_EPSILON:
            h.norm_epsilon = _int_to_norm_eps(value)
        elif key == K_MOE_HIDDEN_DIM:
            h.moe_hidden_dim = value
        else:
            raise ValueError(f"unsupported header key {key}")
    return h.finalize(max_seq_len)


@dataclass
class TensorEntry:
    name: str
    layer: int  # -1 for globals
    expert: int  # -1 for non-expert tensors
    float_type: int
    shape: tuple  # (d, n) for matmuls, (n,) for vectors, (v, dim) for embedding
    offset: int  # byte offset in file
    nbytes: int
    kind: str  # 'full' | 'row' | 'col'  (TP slicing mode)


def tensor_walk(h: LlmHeader) -> list[TensorEntry]:
    """The canonical weight order (reference src/llm.cpp:614-661).

    kind: 'row' = row-split across nodes (output-dim sharded,
    nn-core.cpp:220-230), 'col' = col-split (input-dim sharded,
    nn-core.cpp:232-243), 'full' = replicated on every node.
    """
    entries: list[TensorEntry] = []
    off = h.header_size

    def add(name, layer, expert, ftype, shape, kind):
        nonlocal off
        n_el = int(np.prod(shape))
        nb = tensor_bytes(ftype, n_el)
        entries.append(TensorEntry(name, layer, expert, ftype, tuple(shape), off, nb, kind))
        off += nb

    wt = h.weight_type
    ff = h.ff_dim
    add("embedding", -1, -1, F32, (h.vocab_size, h.dim), "full")
    for l in range(h.n_layers):
        add("block_matmul_q", l, -1, wt, (h.q_dim, h.dim), "row")
        add("block_matmul_k", l, -1, wt, (h.kv_dim, h.dim), "row")
        add("block_matmul_v", l, -1, wt, (h.kv_dim, h.dim), "row")
        add("block_matmul_wo", l, -1, wt, (h.dim, h.q_dim), "col")
        if h.n_experts > 0:
            add("block_moe_gate", l, -1, F32, (h.n_experts, h.dim), "full")
            for e in range(h.n_experts):
                add("block_matmul_w1", l, e, wt, (ff, h.dim), "row")
                add("block_matmul_w2", l, e, wt, (h.dim, ff), "col")
                add("block_matmul_w3", l, e, wt, (ff, h.dim), "row")
        else:
            add("block_matmul_w1", l, -1, wt, (ff, h.dim), "row")
            add("block_matmul_w2", l, -1, wt, (h.dim, ff), "col")
            add("block_matmul_w3", l, -1, wt, (ff, h.dim), "row")
        if h.is_qwen3:
            add("block_norm_q", l, -1, F32, (h.head_dim,), "full")
            add("block_norm_k", l, -1, F32, (h.head_dim,), "full")
        add("block_norm_0", l, -1, F32, (h.dim,), "full")
        add("block_norm_1", l, -1, F32, (h.dim,), "full")
    add("final_norm", -1, -1, F32, (h.dim,), "full")
    add("final_matmul_logits", -1, -1, wt, (h.vocab_size, h.dim), "row")
    return entries


class ModelFile:
    """Memory-mapped .m file with per-tensor, per-rank slice access."""

    def __init__(self, path: str, max_seq_len: int = 0, sync_type: int = Q80):
        self.path = path
        self.header = read_header(path, max_seq_len, sync_type)
        self.entries = tensor_walk(self.header)
        self.by_key = {(e.name, e.layer, e.expert): e for e in self.entries}
        expected = self.entries[-1].offset + self.entries[-1].nbytes
        if expected != self.header.file_size:
            raise ValueError(
                f"weight walk mismatch: expected file size {expected}, "
                f"got {self.header.file_size} (missing {self.header.file_size - expected})")
        self.mm = np.memmap(path, dtype=np.uint8, mode="r")

    def entry(self, name: str, layer: int = -1, expert: int = -1) -> TensorEntry:
        return self.by_key[(name, layer, expert)]

    def raw(self, e: TensorEntry) -> np.ndarray:
        return self.mm[e.offset: e.offset + e.nbytes]

    def f32(self, name: str, layer: int = -1, expert: int = -1) -> np.ndarray:
        """A full f32 tensor (norms, embedding, moe gate)."""
        e = self.entry(name, layer, expert)
        assert e.float_type == F32
        return self.raw(e).view(np.float32).reshape(e.shape)

    def slice_bytes(self, e: TensorEntry, node: int, n_nodes: int) -> np.ndarray:
        """This rank's shard of a matmul weight, as raw block bytes.

        Row split (nn-core.cpp:289-305): contiguous rows
        [node*d0, (node+1)*d0) of the (d, n) weight.
        Col split (nn-core.cpp:307-322): per output row, the byte range of
        columns [node*n0, (node+1)*n0) — block-granular.
        """
        raw = self.raw(e)
        if e.kind == "full" or n_nodes == 1:
            return raw
        d, n = e.shape
        if e.kind == "row":
            assert d % n_nodes == 0
            d0 = d // n_nodes
            row_bytes = tensor_bytes(e.float_type, n)
            return raw[node * d0 * row_bytes: (node + 1) * d0 * row_bytes]
        if e.kind == "col":
            assert n % n_nodes == 0
            n0 = n // n_nodes
            row_bytes = tensor_bytes(e.float_type, n)
            row0_bytes = tensor_bytes(e.float_type, n0)
            rows = raw.reshape(d, row_bytes)
            return np.ascontiguousarray(
                rows[:, node * row0_bytes: (node + 1) * row0_bytes]).reshape(-1)
        raise ValueError(e.kind)

    def slice_q40_planes(self, name: str, layer: int, node: int,
                         n_nodes: int, expert: int = -1):
        """This rank's Q40 shard as device-layout planes (nibble uint8
        [d0, n0/2] + f16 scales [d0, n0/32]) without dequantizing — the
        native CPU matmul streams these directly."""
        e = self.entry(name, layer, expert)
        if e.float_type != Q40:
            raise ValueError(f"{name} is not Q40")
        d, n = e.shape
        d0 = d // n_nodes if e.kind == "row" else d
        n0 = n // n_nodes if e.kind == "col" else n
        raw = self.slice_bytes(e, node, n_nodes)
        return quants.q40_to_planes(raw, d0, n0)

    def slice_f32(self, name: str, layer: int, node: int, n_nodes: int,
                  expert: int = -1) -> np.ndarray:
        """This rank's shard dequantized to f32 with its sliced shape."""
        e = self.entry(name, layer, expert)
        d, n = e.shape
        d0 = d // n_nodes if e.kind == "row" else d
        n0 = n // n_nodes if e.kind == "col" else n
        raw = self.slice_bytes(e, node, n_nodes)
        if e.float_type == F32:
            return raw.view(np.float32).reshape(d0, n0)
        if e.float_type == Q40:
            return quants.dequantize_q40(raw).reshape(d0, n0)
        if e.float_type == Q80:
            return quants.dequantize_q80(raw).reshape(d0, n0)
        raise ValueError(e.float_type)


# ------------------------------------------------------------- writing

def write_header(f, h: LlmHeader) -> None:
    """Serialize a header (converter/writer.py:109-148 semantics)."""
    kv = [
        (K_VERSION, 0),
        (K_ARCH_TYPE, h.arch_type),
        (K_DIM, h.dim),
        (K_HIDDEN_DIM, h.hidden_dim),
        (K_N_LAYERS, h.n_layers),
        (K_N_HEADS, h.n_heads),
        (K_N_KV_HEADS, h.n_kv_heads),
        (K_N_EXPERTS, h.n_experts),
        (K_N_ACTIVE_EXPERTS, h.n_active_experts),
        (K_VOCAB_SIZE, h.vocab_size),
        (K_SEQ_LEN, h.seq_len or h.orig_seq_len),
        (K_HIDDEN_ACT, h.hidden_act),
        (K_ROPE_THETA, int(h.rope_theta)),
        (K_WEIGHT_FLOAT_TYPE, h.weight_type),
        (K_ROPE_TYPE, h.rope_type),
        (K_HEAD_DIM, h.head_dim),
        (K_NORM_EPSILON, _norm_eps_to_int(h.norm_epsilon)),
    ]
    if h.rope_type == ROPE_LLAMA3_1:
        kv += [
            (K_ROPE_SCALING_FACTOR, int(h.rope_scaling_factor)),
            (K_ROPE_SCALING_LOW_FREQ_FACTOR, int(h.rope_scaling_low_freq_factor)),
            (K_ROPE_SCALING_HIGH_FREQ_FACTORY, int(h.rope_scaling_high_freq_factor)),
            (K_ROPE_SCALING_ORIG_MAX_SEQ_LEN, h.rope_scaling_orig_max_seq_len),
        ]
    if h.n_experts > 0:
        kv.append((K_MOE_HIDDEN_DIM, h.moe_hidden_dim))
    data = b"".join(struct.pack("<ii", k, v) for k, v in kv)
    f.write(struct.pack("<ii", MODEL_MAGIC, 8 + len(data)))
    f.write(data)


def write_tensor(f, x: np.ndarray, float_type: int) -> int:
    flat = np.ascontiguousarray(x, dtype=np.float32).reshape(-1)
    if float_type == F32:
        b = flat.tobytes()
    elif float_type == Q40:
        b = quants.quantize_q40(flat).tobytes()
    elif float_type == Q80:
        b = quants.quantize_q80(flat).tobytes()
    else:
        raise ValueError(float_type)
    f.write(b)
    return len(b)


def write_synthetic_model(path: str, h: LlmHeader, seed: int = 1234,
                          scale: float = 0.02, fast: bool = False) -> None:
    """Random-init model in the canonical walk order (for tests/benches;
    there is no network to download real checkpoints).

    fast=True tiles one pre-quantized random block instead of drawing fresh
    values per tensor — content is still well-formed Q40/F32 with sane
    magnitudes, but writing becomes disk-bound (a 40 GB 70B file in tens of
    seconds instead of many minutes of RNG+quantize). Used by the
    big-model load-path harness (tools/load_bigmodel.py)."""
    if h.head_dim == 0:
        h.head_dim = h.dim // h.n_heads
    rng = np.random.default_rng(seed)
    with open(path, "wb") as f:
        write_header(f, h)
    # re-read so header_size/offsets come from the file itself
    hdr = read_header(path)
    pattern: dict[int, bytes] = {}
    if fast:
        import io
        block = rng.standard_normal(size=1 << 22, dtype=np.float32) * scale
        for ft in (F32, Q40):
            buf = io.BytesIO()
            write_tensor(buf, block, ft)
            pattern[ft] = buf.getvalue()
    with open(path, "ab") as f:
        for e in tensor_walk(hdr):
            n = int(np.prod(e.shape))
            if fast:
                pat = pattern[e.float_type]
                want = _tensor_bytes(n, e.float_type)
                reps = -(-want // len(pat))
                f.write((pat * reps)[:want])
            else:
                x = rng.standard_normal(size=n, dtype=np.float32) * scale
                write_tensor(f, x, e.float_type)


def _tensor_bytes(n: int, float_type: int) -> int:
    if float_type == Q40:
        return n // 32 * 18
    return n * 4


# ------------------------------------------------------------- presets

def preset_header(name: str, seq_len: int | None = None) -> LlmHeader:
    """Model-architecture presets used by benches and the sizing audit.

    Shapes follow the public model configs the reference ships converters
    for (reference launch.py:17-73 model registry).
    """
    presets = {
        "llama-3.2-1b": dict(arch_type=ARCH_LLAMA, dim=2048, hidden_dim=8192,
                             n_layers=16, n_heads=32, n_kv_heads=8, head_dim=64,
                             vocab_size=128256, seq_len=131072, rope_theta=500000,
                             rope_type=ROPE_LLAMA3_1, rope_scaling_factor=32,
                             rope_scaling_low_freq_factor=1,
                             rope_scaling_high_freq_factor=4,
                             rope_scaling_orig_max_seq_len=8192),
        "llama-3.2-3b": dict(arch_type=ARCH_LLAMA, dim=3072, hidden_dim=8192,
                             n_layers=28, n_heads=24, n_kv_heads=8, head_dim=128,
                             vocab_size=128256, seq_len=131072, rope_theta=500000,
                             rope_type=ROPE_LLAMA3_1, rope_scaling_factor=32,
                             rope_scaling_low_freq_factor=1,
                             rope_scaling_high_freq_factor=4,
                             rope_scaling_orig_max_seq_len=8192),
        "llama-3.1-8b": dict(arch_type=ARCH_LLAMA, dim=4096, hidden_dim=14336,
                             n_layers=32, n_heads=32, n_kv_heads=8, head_dim=128,
                             vocab_size=128256, seq_len=131072, rope_theta=500000,
                             rope_type=ROPE_LLAMA3_1, rope_scaling_factor=8,
                             rope_scaling_low_freq_factor=1,
                             rope_scaling_high_freq_factor=4,
                             rope_scaling_orig_max_seq_len=8192),
        "llama-3.3-70b": dict(arch_type=ARCH_LLAMA, dim=8192, hidden_dim=28672,
                              n_layers=80, n_heads=64, n_kv_heads=8, head_dim=128,
                              vocab_size=128256, seq_len=131072, rope_theta=500000,
                              rope_type=ROPE_LLAMA3_1, rope_scaling_factor=8,
                              rope_scaling_low_freq_factor=1,
                              rope_scaling_high_freq_factor=4,
                              rope_scaling_orig_max_seq_len=8192),
        "llama-3.1-405b": dict(arch_type=ARCH_LLAMA, dim=16384, hidden_dim=53248,
                               n_layers=126, n_heads=128, n_kv_heads=8, head_dim=128,
                               vocab_size=128256, seq_len=131072, rope_theta=500000,
                               rope_type=ROPE_LLAMA3_1, rope_scaling_factor=8,
                               rope_scaling_low_freq_factor=1,
                               rope_scaling_high_freq_factor=4,
                               rope_scaling_orig_max_seq_len=8192),
        "qwen3-30b-a3b": dict(arch_type=ARCH_QWEN3_MOE, dim=2048, hidden_dim=6144,
                              n_layers=48, n_heads=32, n_kv_heads=4, head_dim=128,
                              n_experts=128, n_active_experts=8, moe_hidden_dim=768,
                              vocab_size=151936, seq_len=40960, rope_theta=1000000,
                              norm_epsilon=1e-6),
        "qwen3-0.6b": dict(arch_type=ARCH_QWEN3, dim=1024, hidden_dim=3072,
                           n_layers=28, n_heads=16, n_kv_heads=8, head_dim=128,
                           vocab_size=151936, seq_len=40960, rope_theta=1000000,
                           norm_epsilon=1e-6),
        "qwen3-1.7b": dict(arch_type=ARCH_QWEN3, dim=2048, hidden_dim=6144,
                           n_layers=28, n_heads=16, n_kv_heads=8, head_dim=128,
                           vocab_size=151936, seq_len=40960, rope_theta=1000000,
                           norm_epsilon=1e-6),
        "qwen3-8b": dict(arch_type=ARCH_QWEN3, dim=4096, hidden_dim=12288,
                         n_layers=36, n_heads=32, n_kv_heads=8, head_dim=128,
                         vocab_size=151936, seq_len=40960, rope_theta=1000000,
                         norm_epsilon=1e-6),
        "qwen3-14b": dict(arch_type=ARCH_QWEN3, dim=5120, hidden_dim=17408,
                          n_layers=40, n_heads=40, n_kv_heads=8, head_dim=128,
                          vocab_size=151936, seq_len=40960, rope_theta=1000000,
                          norm_epsilon=1e-6),
    }
    if name not in presets:
        raise KeyError(f"unknown preset {name}; have {sorted(presets)}")
    h = LlmHeader(**presets[name])
    h.finalize(seq_len or 0)
    return h
