#!/usr/bin/env python3
"""Llama-2 sentencepiece tokenizer.model -> .t converter.

Behavior parity with reference converter/convert-tokenizer-llama2.py:1-44:
  - pieces from SentencePieceProcessor with scores
  - sentencepiece's U+2581 whitespace marker replaced with a plain space
  - the llama2 [INST]/<<SYS>> chat template embedded
  - bos/eos ids from the model

Usage: python converter/convert_tokenizer_llama2.py <folder_with_tokenizer.model> [out.t]
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from dllama_amd.tokenizer import write_tokenizer

# the public Llama-2 chat template (same string the reference embeds,
# convert-tokenizer-llama2.py:6)
CHAT_TEMPLATE = (
    "{% if messages[0]['role'] == 'system' %}{% set loop_messages = messages[1:] %}"
    "{% set system_message = messages[0]['content'] %}{% else %}"
    "{% set loop_messages = messages %}{% set system_message = false %}{% endif %}"
    "{% for message in loop_messages %}"
    "{% if (message['role'] == 'user') != (loop.index0 % 2 == 0) %}"
    "{{ raise_exception('Conversation roles must alternate user/assistant/user/assistant/...') }}"
    "{% endif %}{% if loop.index0 == 0 and system_message != false %}"
    "{% set content = '<<SYS>>\\n' + system_message + '\\n<</SYS>>\\n\\n' + message['content'] %}"
    "{% else %}{% set content = message['content'] %}{% endif %}"
    "{% if message['role'] == 'user' %}{{ bos_token + '[INST] ' + content.strip() + ' [/INST]' }}"
    "{% elif message['role'] == 'assistant' %}{{ ' '  + content.strip() + ' ' + eos_token }}"
    "{% endif %}{% endfor %}")


def convert(model_path: str, out_path: str) -> None:
    from sentencepiece import SentencePieceProcessor
    sp = SentencePieceProcessor(model_file=model_path)
    vocab: list[bytes] = []
    scores: list[float] = []
    for i in range(sp.vocab_size()):
        piece = sp.id_to_piece(i).replace("\u2581", " ")
        b = piece.encode("utf-8")
        if not b:
            b = b"\x00"  # .t requires non-empty pieces
        vocab.append(b)
        scores.append(float(sp.get_score(i)))
    write_tokenizer(out_path, vocab, scores, sp.bos_id(), True,
                    [sp.eos_id()], CHAT_TEMPLATE)
    print(f"✅ Created {out_path} (vocab {len(vocab)}, "
          f"bos {sp.bos_id()}, eos {sp.eos_id()})")


def main() -> int:
    if len(sys.argv) < 2:
        print(__doc__)
        return 1
    src = sys.argv[1]
    if os.path.isdir(src):
        src = os.path.join(src, "tokenizer.model")
    out = sys.argv[2] if len(sys.argv) > 2 else "dllama_tokenizer_llama2.t"
    convert(src, out)
    return 0


if __name__ == "__main__":
    sys.exit(main())
