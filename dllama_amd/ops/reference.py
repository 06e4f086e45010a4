"""PyTorch fp32 reference implementations of every op.

These define the numerics the HIP kernels are tested against (the same role
the f32 CPU kernels play for the quantized kernels in the reference's
nn-cpu-ops-test.cpp:126-277). They also power the CPU backend (BASELINE
config 1 runs without a GPU).

All activation tensors are fp32. Quantized-activation matmuls model the
Q80 round-trip explicitly so CPU and HIP paths see the same quantization
error profile.
"""

from __future__ import annotations

import torch

Q_BLOCK = 32


# ------------------------------------------------------------ quantize

def q80_quantize(x: torch.Tensor):
    """f32 [.., n] -> (q int8 [.., n], s f32 [.., n/32], bsum f32 [.., n/32]).

    d = absmax/127, q = round(x/d) (reference nn-quants.cpp quantizeF32toQ80);
    bsum = sum of the int8 values of the block (used by the GEMV to correct
    for the Q40 nibble offset of 8).
    """
    shape = x.shape
    g = x.reshape(*shape[:-1], shape[-1] // Q_BLOCK, Q_BLOCK).float()
    absmax = g.abs().amax(dim=-1)
    d = absmax / 127.0
    inv = torch.where(d > 0, 1.0 / torch.where(d == 0, torch.ones_like(d), d),
                      torch.zeros_like(d))
    q = torch.round(g * inv.unsqueeze(-1)).clamp(-127, 127).to(torch.int8)
    bsum = q.float().sum(dim=-1)
    return q.reshape(shape), d, bsum


def q80_dequantize(q: torch.Tensor, s: torch.Tensor) -> torch.Tensor:
    shape = q.shape
    g = q.reshape(*shape[:-1], shape[-1] // Q_BLOCK, Q_BLOCK).float()
    return (g * s.unsqueeze(-1)).reshape(shape)


def q80_roundtrip(x: torch.Tensor) -> torch.Tensor:
    q, s, _ = q80_quantize(x)
    return q80_dequantize(q, s)


# ------------------------------------------------------------ matmul

def q40_planes_dequant(qs: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    """Device-layout Q40 planes -> f32 weight [d, n].

    qs uint8 [d, n/2] (block nibbles: byte j of block = elem j | elem j+16<<4),
    scales f16/f32 [d, n/32].
    """
    d, half = qs.shape
    n = half * 2
    nb = n // Q_BLOCK
    b = qs.reshape(d, nb, 16)
    lo = (b & 0xF).to(torch.int8) - 8
    hi = (b >> 4).to(torch.int8) - 8
    vals = torch.cat([lo, hi], dim=-1).float()  # [d, nb, 32]
    vals = vals * scales.float().unsqueeze(-1)
    return vals.reshape(d, n)


def q40_matmul(x: torch.Tensor, w_f32: torch.Tensor,
               quantize_x: bool = True) -> torch.Tensor:
    """y[B, d] = x[B, n] @ w[d, n]^T with the Q80 activation round-trip
    (reference matmul_Q80_Q40_F32, nn-cpu-ops.cpp:231-449)."""
    if quantize_x:
        x = q80_roundtrip(x)
    return x @ w_f32.t()


# ------------------------------------------------------------ norm / act

def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    """y = w * x / sqrt(mean(x^2) + eps) over the last dim
    (reference invRms_F32 + rmsNorm_F32, nn-cpu-ops.cpp:114-175)."""
    inv = torch.rsqrt(x.float().pow(2).mean(dim=-1, keepdim=True) + eps)
    return x * inv * w


def silu(x: torch.Tensor) -> torch.Tensor:
    return x * torch.sigmoid(x)


def swiglu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """silu(a) * b (reference OP_SILU + OP_MUL fusion)."""
    return silu(a) * b


def gelu(x: torch.Tensor) -> torch.Tensor:
    """tanh-approx GELU (reference gelu_F32, nn-cpu-ops.cpp:454-460)."""
    return 0.5 * x * (1.0 + torch.tanh(0.797884560802865 * (x + 0.044715 * x ** 3)))


# ------------------------------------------------------------ rope

def rope_cache(seq_len: int, head_dim: int, theta: float,
               scaling: dict | None = None) -> torch.Tensor:
    """cos/sin cache [seq_len, head_dim/2, 2].

    One head's worth serves every head on every rank: TP slices are whole
    heads, so (global dim % head_dim) == (local dim % head_dim)
    (cf. reference per-node cache, nn-core.cpp:326-383).
    scaling: llama3.1 frequency scaling dict(factor, low_freq_factor,
    high_freq_factor, orig_max_seq_len) (nn-core.cpp:326-340).
    """
    half = head_dim // 2
    j = torch.arange(half, dtype=torch.float32)
    freqs = 1.0 / theta ** (2.0 * j / head_dim)
    if scaling and scaling.get("factor", 1.0) != 1.0:
        freqs = _scale_freqs_llama3(freqs, scaling)
    pos = torch.arange(seq_len, dtype=torch.float32)
    ang = pos[:, None] * freqs[None, :]
    return torch.stack([torch.cos(ang), torch.sin(ang)], dim=-1)


def _scale_freqs_llama3(freqs: torch.Tensor, s: dict) -> torch.Tensor:
    import math
    factor = s["factor"]
    lo_f = s["low_freq_factor"]
    hi_f = s["high_freq_factor"]
    orig = s["orig_max_seq_len"]
    wavelen = 2 * math.pi / freqs
    high_wl = orig / hi_f
    low_wl = orig / lo_f
    smooth = (orig / wavelen - lo_f) / (hi_f - lo_f)
    scaled = torch.where(wavelen < high_wl, freqs,
                         torch.where(wavelen > low_wl, freqs / factor,
                                     (1 - smooth) * freqs / factor + smooth * freqs))
    return scaled


def rope_llama(x: torch.Tensor, cache: torch.Tensor, positions: torch.Tensor,
               head_dim: int) -> torch.Tensor:
    """Interleaved-pair rotation (reference ropeLlama_F32,
    nn-cpu-ops.cpp:843-863). x [B, dim0] with dim0 a multiple of head_dim."""
    B, dim0 = x.shape
    xs = x.reshape(B, dim0 // head_dim, head_dim // 2, 2).float()
    c = cache[positions.long()]  # [B, hd/2, 2]
    cr, ci = c[..., 0].unsqueeze(1), c[..., 1].unsqueeze(1)
    x0, x1 = xs[..., 0], xs[..., 1]
    out = torch.stack([x0 * cr - x1 * ci, x0 * ci + x1 * cr], dim=-1)
    return out.reshape(B, dim0).to(x.dtype)


def rope_falcon(x: torch.Tensor, cache: torch.Tensor, positions: torch.Tensor,
                head_dim: int) -> torch.Tensor:
    """Half-rotated (NeoX) rotation (reference ropeFalcon_F32,
    nn-cpu-ops.cpp:865-885)."""
    B, dim0 = x.shape
    half = head_dim // 2
    xs = x.reshape(B, dim0 // head_dim, 2, half).float()  # [B,H,{lo,hi},half]
    c = cache[positions.long()]
    cr, ci = c[..., 0].unsqueeze(1), c[..., 1].unsqueeze(1)
    x0, x1 = xs[:, :, 0], xs[:, :, 1]
    out = torch.stack([x0 * cr - x1 * ci, x0 * ci + x1 * cr], dim=2)
    return out.reshape(B, dim0).to(x.dtype)


# ------------------------------------------------------------ attention

def attention(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
              positions: torch.Tensor, n_heads0: int, head_dim: int) -> torch.Tensor:
    """Causal decode/prefill attention over the KV cache
    (reference multiheadAtt_F32, nn-cpu-ops.cpp:753-788).

    q [B, n_heads0*head_dim]; k_cache/v_cache [seq, kv_dim0];
    positions [B] — row b attends to cache rows 0..positions[b].
    GQA: kv head = head // (n_heads0*head_dim // kv_dim0 ... ) computed from
    the head ratio.
    """
    B = q.shape[0]
    kv_dim0 = k_cache.shape[1]
    n_kv0 = kv_dim0 // head_dim
    kv_mul = n_heads0 // n_kv0
    scale = 1.0 / head_dim ** 0.5
    qh = q.reshape(B, n_heads0, head_dim).float()
    out = torch.empty_like(qh)
    for b in range(B):
        plen = int(positions[b].item()) + 1
        k = k_cache[:plen].reshape(plen, n_kv0, head_dim).float()
        v = v_cache[:plen].reshape(plen, n_kv0, head_dim).float()
        for h in range(n_heads0):
            kvh = h // kv_mul
            scores = (k[:, kvh] @ qh[b, h]) * scale
            probs = torch.softmax(scores, dim=0)
            out[b, h] = probs @ v[:, kvh]
    return out.reshape(B, n_heads0 * head_dim)


# ------------------------------------------------------------ moe

def moe_gate(router_logits: torch.Tensor, k: int, norm_topk: bool = True):
    """softmax -> top-k -> (normalized) weights
    (reference OP_SOFTMAX + OP_MOE_GATE, nn-cpu-ops.cpp:1443-1492)."""
    probs = torch.softmax(router_logits.float(), dim=-1)
    w, idx = torch.topk(probs, k, dim=-1)
    if norm_topk:
        w = w / w.sum(dim=-1, keepdim=True)
    return idx, w


# ------------------------------------------------------------ sync helpers

def q80_sync_pack(x: torch.Tensor) -> torch.Tensor:
    """Pack an f32 slice [B, n] into the Q80 wire layout used for the
    all-gather sync buffer: int8 payload then f16 scales, per batch row
    (role of reference cast-forward-f32-q80 + SYNC_NODE_SLICES)."""
    B, n = x.shape
    q, s, _ = q80_quantize(x)
    nb = n // Q_BLOCK
    out = torch.empty(B, n + 2 * nb, dtype=torch.uint8)
    out[:, :n] = q.view(torch.uint8)
    out[:, n:] = s.to(torch.float16).view(torch.uint8).reshape(B, 2 * nb)
    return out


def q80_sync_unpack(buf: torch.Tensor, n: int) -> torch.Tensor:
    """Inverse of q80_sync_pack -> f32 [B, n]."""
    B = buf.shape[0]
    nb = n // Q_BLOCK
    q = buf[:, :n].clone().view(torch.int8)
    s = buf[:, n:].contiguous().view(torch.float16).reshape(B, nb)
    return q80_dequantize(q, s.float())
