"""Tokenizer converter parity: sentencepiece (llama2) and tiktoken (llama3)
-> .t (reference converter/convert-tokenizer-llama2.py:1-44 and
convert-tokenizer-llama3.py:1-78). Runs without a GPU."""

import base64
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from dllama_amd.tokenizer import Tokenizer


def test_llama2_sentencepiece_converter(tmp_path):
    spm = pytest.importorskip("sentencepiece")
    corpus = tmp_path / "c.txt"
    corpus.write_text(
        "hello world this is a tiny corpus for the tokenizer test\n" * 50)
    spm.SentencePieceTrainer.train(
        input=str(corpus), model_prefix=str(tmp_path / "tok"),
        vocab_size=64, model_type="bpe", minloglevel=2)
    from converter.convert_tokenizer_llama2 import convert
    out = str(tmp_path / "l2.t")
    convert(str(tmp_path / "tok.model"), out)

    t = Tokenizer(out)
    sp = spm.SentencePieceProcessor(model_file=str(tmp_path / "tok.model"))
    assert len(t.vocab) == sp.vocab_size()
    assert t.bos_id == sp.bos_id()
    assert t.eos_token_ids == [sp.eos_id()]
    assert t.chat_template and "[INST]" in t.chat_template
    # whitespace marker replaced: pieces contain plain spaces
    assert any(p.startswith(b" ") for p in t.vocab)
    # encode/decode round-trips through the greedy-merge BPE
    ids = t.encode("hello world", is_start=True)
    assert ids[0] == t.bos_id
    t.reset_decoder()
    text = "".join(p for p in (t.decode(i) for i in ids[1:]) if p)
    assert text.strip() == "hello world"


def _write_tiktoken(path):
    """Synthesize a full-size llama3-shaped tiktoken file: 256 single-byte
    tokens + unique 3-byte fillers up to 128000 ranks."""
    with open(path, "w") as f:
        for i in range(256):
            f.write(f"{base64.b64encode(bytes([i])).decode()} {i}\n")
        for i in range(256, 128000):
            b = b"\xff" + i.to_bytes(3, "big")
            f.write(f"{base64.b64encode(b).decode()} {i}\n")


def test_llama3_tiktoken_converter(tmp_path):
    src = str(tmp_path / "tokenizer.model")
    _write_tiktoken(src)
    from converter.convert_tokenizer_llama3 import (
        BOS_ID, CHAT_EOS_ID, EOS_ID, convert)
    out = str(tmp_path / "l3.t")
    convert(src, out)

    t = Tokenizer(out)
    assert len(t.vocab) == 128256
    assert t.bos_id == BOS_ID
    assert t.eos_token_ids == [EOS_ID, CHAT_EOS_ID]
    # the special tokens land at their llama3 ids
    assert t.vocab[128000] == b"<|begin_of_text|>"
    assert t.vocab[128006] == b"<|start_header_id|>"
    assert t.vocab[128009] == b"<|eot_id|>"
    assert t.chat_template and "<|start_header_id|>" in t.chat_template
    # special tokens are matched atomically by the encoder
    ids = t.encode("<|eot_id|>", is_start=False)
    assert ids == [128009]
    # ranks became scores: lower rank merges first (score = -rank)
    assert t.scores[0] == 0.0
    assert t.scores[255] == -255.0
