"""Synthetic fixtures: byte-level tokenizer + tiny models for tests/demos
(no network: real checkpoints cannot be downloaded in this environment)."""

from __future__ import annotations

from .. import tokenizer as tok
from .. import model_file as mf


def make_byte_tokenizer(path: str, chat_template: str | None = None) -> None:
    """A byte-level BPE tokenizer: 256 byte tokens, a few merges, and
    llama3-style special tokens. Good enough to exercise the full
    encode -> decode -> chat-template -> eos pipeline."""
    vocab: list[bytes] = [bytes([i]) for i in range(256)]
    scores = [0.0] * 256
    merges = [b"th", b"he", b"the", b" the", b"in", b"an", b"and", b" a",
              b"hello", b" world", b"ll", b"lo", b"wor", b"ld"]
    for i, m in enumerate(merges):
        vocab.append(m)
        scores.append(1.0 + i)  # later merges win ties
    bos_id = len(vocab)
    specials = [b"<|begin_of_text|>", b"<|end_of_text|>", b"<|start_header_id|>",
                b"<|end_header_id|>", b"<|eot_id|>"]
    vocab.extend(specials)
    scores.extend([0.0] * len(specials))
    eos_tokens = [bos_id + 1, bos_id + 4]  # end_of_text, eot_id
    if chat_template is None:
        chat_template = "{{<|start_header_id|>}}"  # auto-detects as llama3
    tok.write_tokenizer(path, vocab, scores, bos_id, True, eos_tokens, chat_template)


def make_tiny_llama(path: str, seed: int = 7, vocab_size: int = 512,
                    seq_len: int = 128, dim: int = 64) -> mf.LlmHeader:
    h = mf.LlmHeader(arch_type=mf.ARCH_LLAMA, dim=dim, hidden_dim=128, n_layers=2,
                     n_heads=4, n_kv_heads=2, head_dim=64, vocab_size=vocab_size,
                     seq_len=seq_len, rope_theta=10000, rope_type=mf.ROPE_LLAMA)
    h.finalize()
    mf.write_synthetic_model(path, h, seed=seed)
    return h


def make_tiny_qwen3(path: str, seed: int = 9, moe: bool = False,
                    dim: int = 64) -> mf.LlmHeader:
    if moe:
        h = mf.LlmHeader(arch_type=mf.ARCH_QWEN3_MOE, dim=dim, hidden_dim=96,
                         n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                         n_experts=4, n_active_experts=2, moe_hidden_dim=64,
                         vocab_size=256, seq_len=64, rope_theta=10000,
                         norm_epsilon=1e-6)
    else:
        h = mf.LlmHeader(arch_type=mf.ARCH_QWEN3, dim=64, hidden_dim=96,
                         n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                         vocab_size=256, seq_len=64, rope_theta=10000,
                         norm_epsilon=1e-6)
    h.finalize()
    mf.write_synthetic_model(path, h, seed=seed)
    return h
