"""Quantization round-trip tests.

Mirrors the reference test strategy (src/nn/nn-cpu-ops-test.cpp:87-104):
Q40/Q80 quantize<->dequantize round-trips with the same tolerances
(0.13 absolute for Q40 on [-1,1]-ish data scaled, 0.01 for Q80).
"""

import numpy as np
import pytest

from dllama_amd import quants


def test_q80_roundtrip():
    rng = np.random.default_rng(0)
    x = (rng.standard_normal(4096).astype(np.float32)) * 2
    y = quants.q80_roundtrip(x)
    # reference tolerance: 0.01 on k/nBlocks-scaled ramp; scale-relative here
    assert np.abs(x - y).max() <= np.abs(x).max() / 127.0 * 1.01


def test_q40_roundtrip():
    rng = np.random.default_rng(1)
    x = rng.standard_normal(4096).astype(np.float32)
    y = quants.q40_roundtrip(x)
    # Q40 is 4-bit: block-absmax/8 max error plus rounding slack
    assert np.abs(x - y).max() <= np.abs(x).max() / 8.0 * 1.2


def test_q40_block_layout():
    # byte j of a block must hold elem j (lo nibble) and elem j+16 (hi nibble)
    x = np.arange(32, dtype=np.float32) - 16
    b = quants.quantize_q40(x)
    assert b.shape == (1, 18)
    d = b[0, :2].copy().view(np.float16).astype(np.float32)[0]
    lo = (b[0, 2:] & 0xF).astype(np.int32) - 8
    hi = ((b[0, 2:] >> 4) & 0xF).astype(np.int32) - 8
    recon = np.concatenate([lo, hi]) * d
    assert np.abs(recon - x).max() <= np.abs(x).max() / 8.0 * 1.2


def test_q80_zeros():
    x = np.zeros(64, dtype=np.float32)
    y = quants.q80_roundtrip(x)
    assert np.all(y == 0)


def test_q40_planes():
    rng = np.random.default_rng(2)
    d_rows, n_cols = 8, 96
    w = rng.standard_normal((d_rows, n_cols)).astype(np.float32)
    blocks = quants.quantize_q40(w)
    qs, scales = quants.q40_to_planes(blocks, d_rows, n_cols)
    assert qs.shape == (d_rows, n_cols // 2)
    assert scales.shape == (d_rows, n_cols // 32)
    # reconstruct from planes and compare to block dequant
    ref = quants.dequantize_q40(blocks, d_rows * n_cols).reshape(d_rows, n_cols)
    lo = (qs & 0xF).astype(np.int8) - 8
    hi = (qs >> 4).astype(np.int8) - 8
    nb = n_cols // 32
    vals = np.empty((d_rows, n_cols), dtype=np.float32)
    for b in range(nb):
        vals[:, b * 32: b * 32 + 16] = lo[:, b * 16:(b + 1) * 16]
        vals[:, b * 32 + 16:(b + 1) * 32] = hi[:, b * 16:(b + 1) * 16]
    vals *= np.repeat(scales.astype(np.float32), 32, axis=1)
    assert np.allclose(vals, ref)


def test_quantizers_byte_exact_vs_reference_converter(tmp_path):
    """Run the reference's OWN Python converter (read-only import from
    /root/reference/converter/writer.py) on the same data and require
    byte-identical Q40 and Q80 output — pins the wire format beyond the
    layout unit tests."""
    import importlib.util
    import io
    import os

    import numpy as np
    import pytest

    path = "/root/reference/converter/writer.py"
    if not os.path.exists(path):
        pytest.skip("reference converter not available")
    spec = importlib.util.spec_from_file_location("ref_writer", path)
    ref = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(ref)
    except Exception as e:  # noqa: BLE001
        pytest.skip(f"reference writer import failed: {e}")

    rng = np.random.default_rng(17)
    x = (rng.standard_normal(2048) * 0.7).astype(np.float32)
    x[:32] = 0.0  # all-zero block edge case

    import torch
    buf = io.BytesIO()
    with np.errstate(divide="ignore"):  # the reference writer divides by 0
        ref.writeQuantizedQ40Tensor(buf, torch.from_numpy(x))
    assert buf.getvalue() == quants.quantize_q40(x).tobytes()

    buf = io.BytesIO()
    with np.errstate(divide="ignore"):
        ref.writeQuantizedQ80Tensor(buf, torch.from_numpy(x))
    assert buf.getvalue() == quants.quantize_q80(x).tobytes()
