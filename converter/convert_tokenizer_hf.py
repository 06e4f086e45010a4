#!/usr/bin/env python3
"""HF tokenizer -> .t converter.

Behavior parity with reference converter/convert-tokenizer-hf.py:
  - byte-level BPE vocab decoded through the GPT-2 unicode->byte table
    (reference convert-tokenizer-hf.py:12-23)
  - scores = -token_id for HF fast tokenizers (:47)
  - bos/eos from tokenizer or config.json; chat template embedded.

Usage: python converter/convert_tokenizer_hf.py <hf_folder> <name>
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from dllama_amd.tokenizer import write_tokenizer


def unicode_to_bytes() -> dict:
    """GPT-2 byte-level BPE unicode escape table (openai/gpt-2 encoder.py)."""
    bs = (list(range(ord("!"), ord("~") + 1))
          + list(range(ord("¡"), ord("¬") + 1))
          + list(range(ord("®"), ord("ÿ") + 1)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return {chr(c): b for c, b in zip(cs, bs)}


def token_to_bytes(token: str, utb: dict) -> bytes:
    out = bytearray()
    for ch in token:
        if ch in utb:
            out.append(utb[ch])
        else:
            out.extend(ch.encode("utf-8"))
    return bytes(out)


def convert(folder: str, out_path: str) -> None:
    with open(os.path.join(folder, "tokenizer_config.json")) as f:
        tok_cfg = json.load(f)
    cls = tok_cfg.get("tokenizer_class", "PreTrainedTokenizerFast")
    utb = unicode_to_bytes()

    if cls in ("PreTrainedTokenizerFast", "LlamaTokenizerFast", "Qwen2Tokenizer"):
        from transformers import PreTrainedTokenizerFast
        tk = PreTrainedTokenizerFast(
            tokenizer_file=os.path.join(folder, "tokenizer.json"))
        vocab_len = len(tk.get_vocab())
        tokens = [token_to_bytes(tk.convert_ids_to_tokens([i])[0], utb)
                  for i in range(vocab_len)]
        scores = [-float(i) for i in range(vocab_len)]
        bos_id = tk.bos_token_id
        eos_ids = [tk.eos_token_id] if tk.eos_token_id is not None else None
    elif cls == "LlamaTokenizer":
        from sentencepiece import SentencePieceProcessor
        sp = SentencePieceProcessor(
            model_file=os.path.join(folder, "tokenizer.model"))
        bos_id = sp.bos_id()
        eos_ids = [sp.eos_id()]
        tokens, scores = [], []
        for i in range(sp.vocab_size()):
            t = sp.id_to_piece(i).replace("▁", " ")
            if len(t) == 6 and t.startswith("<0x") and t.endswith(">"):
                b = bytes.fromhex(t[3:-1])
            else:
                b = t.encode("utf-8")
            tokens.append(b)
            scores.append(sp.get_score(i))
    else:
        raise ValueError(f"tokenizer class {cls} not supported")

    if bos_id is None or eos_ids is None:
        with open(os.path.join(folder, "config.json")) as f:
            cfg = json.load(f)
        bos_id = bos_id if bos_id is not None else cfg["bos_token_id"]
        if eos_ids is None:
            e = cfg["eos_token_id"]
            eos_ids = e if isinstance(e, list) else [e]

    chat_template = tok_cfg.get("chat_template")
    add_bos = bool(tok_cfg.get("add_bos_token", True))
    write_tokenizer(out_path, tokens, scores, bos_id, add_bos, eos_ids,
                    chat_template)
    print(f"✅ {out_path} created (vocab {len(tokens)}, bos {bos_id}, eos {eos_ids})")


def main():
    if len(sys.argv) < 3:
        print(__doc__)
        return 1
    convert(sys.argv[1], f"dllama_tokenizer_{sys.argv[2]}.t")
    return 0


if __name__ == "__main__":
    sys.exit(main())
