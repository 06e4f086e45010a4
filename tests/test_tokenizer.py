"""Tokenizer tests, mirroring the reference tokenizer-test strategy
(src/tokenizer-test.cpp): chat-template auto-detection, EosDetector
streaming state machine, encode/decode round trips, UTF-8 stream recovery."""

import numpy as np
import pytest

from dllama_amd import tokenizer as tok
from dllama_amd.utils.testing import make_byte_tokenizer


@pytest.fixture(scope="module")
def byte_tok(tmp_path_factory):
    path = str(tmp_path_factory.mktemp("t") / "byte.t")
    make_byte_tokenizer(path)
    return tok.Tokenizer(path)


def test_header(byte_tok):
    assert byte_tok.bos_id == 256 + 14
    assert byte_tok.add_bos
    assert len(byte_tok.eos_token_ids) == 2
    assert byte_tok.chat_template is not None


def test_encode_merges(byte_tok):
    ids = byte_tok.encode("hello", is_start=False)
    # greedy pair-merging compresses the 5 bytes (he + l + lo)
    assert len(ids) < 5
    assert b"".join(byte_tok.vocab[i] for i in ids) == b"hello"


def test_encode_bos(byte_tok):
    ids = byte_tok.encode("hi", is_start=True)
    assert ids[0] == byte_tok.bos_id


def test_encode_special_tokens(byte_tok):
    ids = byte_tok.encode("<|start_header_id|>user<|end_header_id|>", is_start=False)
    assert ids[0] == byte_tok.bos_id + 2
    assert ids[-1] == byte_tok.bos_id + 3


def test_decode_roundtrip(byte_tok):
    text = "the world and a hello"
    ids = byte_tok.encode(text, is_start=False)
    byte_tok.reset_decoder()
    out = "".join(p for p in (byte_tok.decode(t) for t in ids) if p)
    assert out == text


def test_decode_utf8_streaming(byte_tok):
    # multi-byte emoji split across byte tokens must buffer until complete
    emoji = "🙃".encode("utf-8")  # 4 bytes
    byte_tok.reset_decoder()
    outs = [byte_tok.decode(b) for b in emoji]
    assert outs[:3] == [None, None, None]
    assert outs[3] == "🙃"


def test_template_autodetect():
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_UNKNOWN, "...<|start_header_id|>...", "<eos>")
    assert g.type == tok.TEMPLATE_LLAMA3
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_UNKNOWN, "xx [INST] yy", "<eos>")
    assert g.type == tok.TEMPLATE_LLAMA2
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_UNKNOWN, "a<|im_start|>b", "<eos>")
    assert g.type == tok.TEMPLATE_CHATML
    with pytest.raises(ValueError):
        tok.ChatTemplateGenerator(tok.TEMPLATE_UNKNOWN, "nothing", "<eos>")


def test_template_llama3_output():
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_LLAMA3, None, "<|eot_id|>")
    out = g.generate([tok.ChatItem("user", "hi")], True)
    assert out.content == ("<|start_header_id|>user<|end_header_id|>\n\nhi<|eot_id|>"
                           "<|start_header_id|>assistant<|end_header_id|>\n\n")


def test_template_llama2_output():
    """Golden vs reference tokenizer.cpp:578-591 ([INST]/<<SYS>> fusion of
    a leading system+user pair, eos after every turn)."""
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_LLAMA2, None, "</s>")
    out = g.generate([tok.ChatItem("system", "S"),
                      tok.ChatItem("user", "U1"),
                      tok.ChatItem("assistant", "A1"),
                      tok.ChatItem("user", "U2")], True)
    assert out.content == ("[INST] <<SYS>>\nS\n<</SYS>>\n\nU1 [/INST]</s>"
                           "A1</s>[INST] U2 [/INST]</s>")


def test_template_deepseek3_output():
    """Golden vs reference tokenizer.cpp:597-614: bare system prefix,
    <think> generation prompt whose public part is the trailing 8 bytes."""
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_DEEP_SEEK3, None, "<eos>")
    out = g.generate([tok.ChatItem("system", "S"),
                      tok.ChatItem("user", "U")], True)
    assert out.content == "S<\uff5cUser\uff5c>U<\uff5cAssistant\uff5c><think>\n"
    assert out.public_prompt == "<think>\n"


def test_template_chatml_output():
    """Golden vs reference tokenizer.cpp:615-627 INCLUDING its quirk: the
    generation prompt is appended inside the per-item loop (once per
    message), kept for exact parity."""
    g = tok.ChatTemplateGenerator(tok.TEMPLATE_CHATML, None, "<eos>")
    out = g.generate([tok.ChatItem("user", "U")], True)
    assert out.content == "<|im_start|>user\nU<|im_end|>\n<|im_start|>assistant\n"
    out2 = g.generate([tok.ChatItem("user", "U"),
                       tok.ChatItem("assistant", "A")], True)
    assert out2.content == ("<|im_start|>user\nU<|im_end|>\n<|im_start|>assistant\n"
                            "<|im_start|>assistant\nA<|im_end|>\n<|im_start|>assistant\n")


def test_eos_detector_exact():
    d = tok.EosDetector([99], ["<stop>"], 0, 0)
    assert d.append(1, "<sto") == tok.MAYBE_EOS
    assert d.append(2, "p>") == tok.EOS
    assert d.get_delta() is None


def test_eos_detector_token_id():
    d = tok.EosDetector([99], ["<stop>"], 0, 0)
    assert d.append(99, None) == tok.EOS


def test_eos_detector_not_eos():
    d = tok.EosDetector([99], ["<stop>"], 0, 0)
    assert d.append(1, "hello ") == tok.NOT_EOS
    assert d.get_delta() == "hello "
    d.reset()
    assert d.get_delta() is None


def test_eos_detector_padding():
    # reference tokenizer-test.cpp exercises left/right padded stops
    d = tok.EosDetector([99], ["</s>"], 2, 2)
    assert d.append(1, "x</s") == tok.MAYBE_EOS
    assert d.append(2, ">") == tok.EOS
    assert d.get_delta() == "x"


def test_sampler_greedy():
    s = tok.Sampler(10, 0.0, 0.9, 123)
    logits = np.zeros(10, dtype=np.float32)
    logits[7] = 5.0
    assert s.sample(logits) == 7


def test_sampler_topp_distribution():
    s = tok.Sampler(4, 1.0, 0.5, 42)
    logits = np.array([10.0, 1.0, 0.5, 0.1], dtype=np.float32)
    counts = np.zeros(4)
    for _ in range(100):
        counts[s.sample(logits)] += 1
    assert counts[0] == 100  # top-p 0.5 keeps only the dominant token


def test_sampler_deterministic_seed():
    a = tok.Sampler(100, 0.8, 0.9, 7)
    b = tok.Sampler(100, 0.8, 0.9, 7)
    logits = np.random.default_rng(0).standard_normal(100).astype(np.float32)
    assert [a.sample(logits) for _ in range(10)] == [b.sample(logits) for _ in range(10)]


def test_sampler_torch_path_matches_numpy():
    """sample_torch (the GPU serving path — works on CPU tensors too) must
    track the numpy reference path: identical greedy picks, and identical
    sampled tokens for the same xorshift coin on a well-separated
    distribution (boundary-exact ties are measure-zero and excluded by
    construction)."""
    import torch
    import numpy as np
    logits = np.zeros(64, dtype=np.float32)
    logits[7], logits[21], logits[42] = 8.0, 6.0, 4.0  # dominant trio
    t = torch.tensor(logits)

    g1 = tok.Sampler(64, 0.0, 0.9, 3)
    g2 = tok.Sampler(64, 0.0, 0.9, 3)
    assert g1.sample(logits) == g2.sample_torch(t) == 7

    for topp in (0.95, 1.0):
        s1 = tok.Sampler(64, 1.0, topp, 1234)
        s2 = tok.Sampler(64, 1.0, topp, 1234)
        picks1 = [s1.sample(logits) for _ in range(100)]
        picks2 = [s2.sample_torch(t) for _ in range(100)]
        assert picks1 == picks2, (topp, picks1[:5], picks2[:5])
        # the dominant trio carries ~98% of the mass
        assert sum(p in (7, 21, 42) for p in picks1) >= 90


def test_tokenizer_writer_byte_exact_vs_reference_converter(tmp_path):
    """Our .t writer must produce byte-identical files to the reference's
    own converter (converter/tokenizer-writer.py) for the same vocab."""
    import importlib.util
    import io
    import os

    path = "/root/reference/converter/tokenizer-writer.py"
    if not os.path.exists(path):
        pytest.skip("reference converter not available")
    spec = importlib.util.spec_from_file_location("ref_tok_writer", path)
    ref = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(ref)

    vocab = [b"a", b"b", b"ab", b"<|eot|>"]
    scores = [0.0, 0.5, 1.0, 0.0]
    template = b"{{x [INST] y}}"
    buf = io.BytesIO()
    ref.writeTokenizer(buf, vocab, scores, template, 3, True, [3])

    ours = str(tmp_path / "t.t")
    tok.write_tokenizer(ours, vocab, scores, 3, True, [3],
                        template.decode())
    with open(ours, "rb") as f:
        assert f.read() == buf.getvalue()
