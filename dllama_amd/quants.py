"""Q40 / Q80 block quantization.

Format parity with the reference (b4rtaz/distributed-llama):
  - block size 32 (reference src/nn/nn-quants.hpp:53-54)
  - Q40 block = f16 scale `d` + 16 bytes of packed nibbles
    (reference src/nn/nn-quants.hpp:64-67): byte j holds element j in its
    low nibble and element j+16 in its high nibble; value = (nib - 8) * d.
  - Q80 block = f16 scale `d` + 32 int8 values (nn-quants.hpp:69-72);
    value = q * d.
  - quantization rules follow the reference converter
    (converter/writer.py:29-74): Q40 d = max-abs-signed / -8 with the
    +8.5 offset trick; Q80 d = absmax / 127.

Everything here is vectorized numpy; torch helpers convert to the device
layout used by the HIP kernels (nibble plane + f16 scale plane).
"""

from __future__ import annotations

import numpy as np

Q_BLOCK = 32  # elements per block (Q40 and Q80)
Q40_BLOCK_BYTES = 18  # 2 (f16 d) + 16 (nibbles)
Q80_BLOCK_BYTES = 34  # 2 (f16 d) + 32 (int8)

# float type ids used in .m headers (reference src/nn/nn-quants.hpp:56-62)
F32 = 0
F16 = 1
Q40 = 2
Q80 = 3

_FLOAT_NAMES = {F32: "f32", F16: "f16", Q40: "q40", Q80: "q80"}


def float_type_name(t: int) -> str:
    return _FLOAT_NAMES.get(t, f"unk({t})")


def tensor_bytes(float_type: int, n_elements: int) -> int:
    """Size in bytes of a flat tensor of `n_elements` in the given format
    (reference src/nn/nn-core.cpp getBytes)."""
    if float_type == F32:
        return 4 * n_elements
    if float_type == F16:
        return 2 * n_elements
    if float_type == Q40:
        assert n_elements % Q_BLOCK == 0
        return n_elements // Q_BLOCK * Q40_BLOCK_BYTES
    if float_type == Q80:
        assert n_elements % Q_BLOCK == 0
        return n_elements // Q_BLOCK * Q80_BLOCK_BYTES
    raise ValueError(f"unsupported float type {float_type}")


# ---------------------------------------------------------------- Q40

def quantize_q40(x: np.ndarray) -> np.ndarray:
    """f32 -> Q40 blocks. Returns uint8 array [nblocks, 18].

    Mirrors the reference converter math (converter/writer.py:29-53):
    d = (signed value with largest magnitude) / -8, q = clip(x/d + 8.5, 0, 15)
    """
    x = np.ascontiguousarray(x, dtype=np.float32).reshape(-1)
    assert x.size % Q_BLOCK == 0, x.size
    g = x.reshape(-1, Q_BLOCK)
    gmax = g.max(axis=1)
    gmin = g.min(axis=1)
    d = np.where(-gmin > gmax, gmin, gmax) / -8.0
    d16 = d.astype(np.float16)
    inv = np.where(d != 0, 1.0 / np.where(d == 0, 1.0, d), 0.0)
    q = np.clip(g * inv[:, None] + 8.5, 0, 15).astype(np.uint8)
    lo = q[:, : Q_BLOCK // 2] & 0xF
    hi = (q[:, Q_BLOCK // 2:] & 0xF) << 4
    packed = lo | hi
    out = np.empty((g.shape[0], Q40_BLOCK_BYTES), dtype=np.uint8)
    out[:, :2] = d16.view(np.uint8).reshape(-1, 2)
    out[:, 2:] = packed
    return out


def dequantize_q40(blocks: np.ndarray, n: int | None = None) -> np.ndarray:
    """Q40 blocks [nblocks, 18] (or flat bytes) -> f32 flat array."""
    b = np.ascontiguousarray(blocks, dtype=np.uint8).reshape(-1, Q40_BLOCK_BYTES)
    d = b[:, :2].copy().view(np.float16).astype(np.float32).reshape(-1)
    qs = b[:, 2:]
    lo = (qs & 0xF).astype(np.int8) - 8
    hi = (qs >> 4).astype(np.int8) - 8
    vals = np.concatenate([lo, hi], axis=1).astype(np.float32) * d[:, None]
    out = vals.reshape(-1)
    if n is not None:
        out = out[:n]
    return out


def q40_to_planes(blocks: np.ndarray, d_rows: int, n_cols: int):
    """Q40 blocks of a row-major (d_rows, n_cols) weight -> device layout:

    qs_plane  uint8 [d_rows, n_cols/2] — raw 16-byte nibble payloads,
              blocks of a row contiguous (same packing as the wire format:
              byte j of block = elem j | elem j+16 << 4)
    scales    float16 [d_rows, n_cols/32]

    The HIP GEMV reads qs as uint4 (16B = 1 block) per lane.
    """
    b = np.ascontiguousarray(blocks, dtype=np.uint8).reshape(-1, Q40_BLOCK_BYTES)
    nblocks_per_row = n_cols // Q_BLOCK
    assert b.shape[0] == d_rows * nblocks_per_row, (b.shape, d_rows, n_cols)
    scales = b[:, :2].copy().view(np.float16).reshape(d_rows, nblocks_per_row)
    qs = b[:, 2:].reshape(d_rows, nblocks_per_row * (Q_BLOCK // 2))
    return np.ascontiguousarray(qs), np.ascontiguousarray(scales)


# ---------------------------------------------------------------- Q80

def quantize_q80(x: np.ndarray) -> np.ndarray:
    """f32 -> Q80 blocks. Returns uint8 array [nblocks, 34].

    Mirrors converter/writer.py:55-74: d = absmax/127, q = round(x/d).
    """
    x = np.ascontiguousarray(x, dtype=np.float32).reshape(-1)
    assert x.size % Q_BLOCK == 0
    g = x.reshape(-1, Q_BLOCK)
    absmax = np.abs(g).max(axis=1)
    d = absmax / 127.0
    d16 = d.astype(np.float16)
    inv = np.where(d != 0, 1.0 / np.where(d == 0, 1.0, d), 0.0)
    q = np.round(g * inv[:, None]).astype(np.int8)
    out = np.empty((g.shape[0], Q80_BLOCK_BYTES), dtype=np.uint8)
    out[:, :2] = d16.view(np.uint8).reshape(-1, 2)
    out[:, 2:] = q.view(np.uint8)
    return out


def dequantize_q80(blocks: np.ndarray, n: int | None = None) -> np.ndarray:
    b = np.ascontiguousarray(blocks, dtype=np.uint8).reshape(-1, Q80_BLOCK_BYTES)
    d = b[:, :2].copy().view(np.float16).astype(np.float32).reshape(-1)
    q = b[:, 2:].copy().view(np.int8).astype(np.float32)
    out = (q * d[:, None]).reshape(-1)
    if n is not None:
        out = out[:n]
    return out


def q80_roundtrip(x: np.ndarray) -> np.ndarray:
    """Quantize-dequantize in one step (the numeric effect of a Q80 cast)."""
    return dequantize_q80(quantize_q80(x), x.size).reshape(x.shape)


def q40_roundtrip(x: np.ndarray) -> np.ndarray:
    return dequantize_q40(quantize_q40(x), x.size).reshape(x.shape)
