"""`.m` model format tests: header round-trip, walk integrity, TP slicing.

Slicing parity targets reference nn-core.cpp:220-243 (row/col slices) and
nn-core.cpp:289-322 (weight splitters): stacking all ranks' shards must
reproduce the full tensor.
"""

import numpy as np
import pytest

from dllama_amd import model_file as mf
from dllama_amd import quants


@pytest.fixture(scope="module")
def tiny_model(tmp_path_factory):
    path = str(tmp_path_factory.mktemp("m") / "tiny.m")
    h = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=64, hidden_dim=128, n_layers=2,
                     n_heads=4, n_kv_heads=2, head_dim=16, vocab_size=256,
                     seq_len=128, rope_theta=10000, rope_type=mf.ROPE_LLAMA)
    h.finalize()
    mf.write_synthetic_model(path, h, seed=7)
    return path


def test_header_roundtrip(tiny_model):
    h = mf.read_header(tiny_model)
    assert h.dim == 64
    assert h.n_layers == 2
    assert h.q_dim == 64
    assert h.kv_dim == 32
    assert h.weight_type == quants.Q40
    assert h.norm_epsilon == pytest.approx(1e-5)


def test_walk_covers_file(tiny_model):
    m = mf.ModelFile(tiny_model)  # ctor asserts walk == file size
    names = {e.name for e in m.entries}
    assert {"embedding", "block_matmul_q", "block_matmul_wo", "final_norm",
            "final_matmul_logits"} <= names


def test_row_col_slices(tiny_model):
    m = mf.ModelFile(tiny_model)
    for n_nodes in (1, 2):
        full_q = m.slice_f32("block_matmul_q", 0, 0, 1)
        parts = [m.slice_f32("block_matmul_q", 0, r, n_nodes) for r in range(n_nodes)]
        assert np.allclose(np.vstack(parts), full_q)
        full_wo = m.slice_f32("block_matmul_wo", 0, 0, 1)
        parts = [m.slice_f32("block_matmul_wo", 0, r, n_nodes) for r in range(n_nodes)]
        assert np.allclose(np.hstack(parts), full_wo)


def test_seq_len_clamp(tiny_model):
    h = mf.read_header(tiny_model, max_seq_len=16)
    assert h.seq_len == 16
    assert h.orig_seq_len == 128


def test_moe_walk():
    h = mf.LlmHeader(arch_type=mf.ARCH_QWEN3_MOE, dim=64, hidden_dim=96,
                     n_layers=1, n_heads=4, n_kv_heads=2, head_dim=16,
                     n_experts=4, n_active_experts=2, moe_hidden_dim=32,
                     vocab_size=128, seq_len=64, rope_theta=10000,
                     norm_epsilon=1e-6)
    h.finalize()
    entries = None
    h.header_size = 100  # arbitrary; offsets are relative
    entries = mf.tensor_walk(h)
    names = [(e.name, e.expert) for e in entries]
    assert ("block_moe_gate", -1) in names
    assert ("block_matmul_w1", 0) in names
    assert ("block_matmul_w2", 3) in names
    assert ("block_norm_q", -1) in names  # qwen3 qk-norm present
    assert h.rope_type == mf.ROPE_FALCON  # qwen3 forces falcon rope


def test_bad_magic_and_truncation(tmp_path):
    """Corrupt .m files produce clear errors, not garbage loads (reference
    llm.cpp:36-116 asserts the magic and walks to EOF)."""
    import pytest
    from dllama_amd.utils.testing import make_tiny_llama
    bad = tmp_path / "bad.m"
    bad.write_bytes(b"\x00" * 64)
    with pytest.raises(ValueError, match="magic"):
        mf.read_header(str(bad))

    p = tmp_path / "trunc.m"
    make_tiny_llama(str(p))
    data = p.read_bytes()
    p.write_bytes(data[: len(data) - 100])
    with pytest.raises(ValueError, match="missing"):
        mf.ModelFile(str(p))


def test_unknown_preset():
    import pytest
    with pytest.raises(KeyError, match="unknown preset"):
        mf.preset_header("llama-9000b")


def test_write_synthetic_fast_matches_layout(tmp_path):
    """fast=True produces a byte-layout-identical file structure (same
    header, same tensor byte extents) as the slow writer."""
    from dllama_amd import model_file as mf
    h1 = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=64, hidden_dim=128,
                      n_layers=2, n_heads=4, n_kv_heads=2, head_dim=16,
                      vocab_size=128, seq_len=64, rope_theta=10000,
                      rope_type=mf.ROPE_LLAMA)
    h1.finalize()
    p_slow = str(tmp_path / "slow.m")
    p_fast = str(tmp_path / "fast.m")
    mf.write_synthetic_model(p_slow, h1, seed=3)
    h2 = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=64, hidden_dim=128,
                      n_layers=2, n_heads=4, n_kv_heads=2, head_dim=16,
                      vocab_size=128, seq_len=64, rope_theta=10000,
                      rope_type=mf.ROPE_LLAMA)
    h2.finalize()
    mf.write_synthetic_model(p_fast, h2, seed=3, fast=True)
    import os
    assert os.path.getsize(p_fast) == os.path.getsize(p_slow)
    # both parse and walk identically
    a = mf.ModelFile(p_slow)
    b = mf.ModelFile(p_fast)
    for ea, eb in zip(mf.tensor_walk(a.header), mf.tensor_walk(b.header)):
        assert ea.name == eb.name and ea.shape == eb.shape
        assert ea.offset == eb.offset
