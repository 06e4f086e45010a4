#!/usr/bin/env python3
"""Llama-3 tiktoken tokenizer.model -> .t converter.

Behavior parity with reference converter/convert-tokenizer-llama3.py:1-78:
  - input is the tiktoken format: one `<base64-bytes> <rank>` pair per line
  - scores are the negated ranks (BPE merge priority for the greedy encoder)
  - the 256 llama-3 special tokens appended after the base vocab with
    descending scores
  - bos 128000; eos = [128001 end_of_text, 128009 eot_id]; llama3 chat
    template embedded

Usage: python converter/convert_tokenizer_llama3.py <tokenizer.model> [out.t]
"""

import base64
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from dllama_amd.tokenizer import write_tokenizer

N_SPECIAL = 256
SPECIAL_TOKENS = [
    "<|begin_of_text|>",
    "<|end_of_text|>",
    "<|reserved_special_token_0|>",
    "<|reserved_special_token_1|>",
    "<|reserved_special_token_2|>",
    "<|reserved_special_token_3|>",
    "<|start_header_id|>",
    "<|end_header_id|>",
    "<|reserved_special_token_4|>",
    "<|eot_id|>",
] + [f"<|reserved_special_token_{i}|>" for i in range(5, N_SPECIAL - 5)]

BOS_ID = 128000
EOS_ID = 128001
CHAT_EOS_ID = 128009

# the public Llama-3 chat template (reference convert-tokenizer-llama3.py:33)
CHAT_TEMPLATE = (
    "{% set loop_messages = messages %}{% for message in loop_messages %}"
    "{% set content = '<|start_header_id|>' + message['role'] + '<|end_header_id|>\n\n'"
    "+ message['content'] | trim + '<|eot_id|>' %}"
    "{% if loop.index0 == 0 %}{% set content = bos_token + content %}{% endif %}"
    "{{ content }}{% endfor %}{% if add_generation_prompt %}"
    "{{ '<|start_header_id|>assistant<|end_header_id|>\n\n' }}{% endif %}")


def convert(model_path: str, out_path: str) -> None:
    vocab: list[bytes] = []
    scores: list[float] = []
    with open(model_path) as f:
        for line in f:
            if not line.strip():
                continue
            b64, rank = line.split()
            vocab.append(base64.b64decode(b64))
            scores.append(-float(rank))
    base = len(vocab)
    for i, tok in enumerate(SPECIAL_TOKENS):
        vocab.append(tok.encode("utf-8"))
        scores.append(-float(base + i))
    write_tokenizer(out_path, vocab, scores, BOS_ID, True,
                    [EOS_ID, CHAT_EOS_ID], CHAT_TEMPLATE)
    print(f"✅ Created {out_path} (vocab {len(vocab)})")


def main() -> int:
    if len(sys.argv) < 2:
        print(__doc__)
        return 1
    src = sys.argv[1]
    if os.path.isdir(src):
        src = os.path.join(src, "tokenizer.model")
    out = sys.argv[2] if len(sys.argv) > 2 else "dllama_tokenizer_llama3.t"
    convert(src, out)
    return 0


if __name__ == "__main__":
    sys.exit(main())
