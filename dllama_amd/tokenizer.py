"""`.t` tokenizer format, BPE encode/decode, sampler, chat templates, EOS detector.

Format and behavior parity with the reference:
  - .t layout: magic 0x567124, KV header, optional chat template, eos id list,
    then vocab entries (f32 score, i32 length, bytes)
    — reference src/tokenizer.cpp:42-164, converter/tokenizer-writer.py:3-57.
  - encode: special-token scan + exact-match byte accumulation, then greedy
    highest-score pair merging — reference src/tokenizer.cpp:311-390.
  - streaming decode with UTF-8 recovery — reference src/tokenizer.cpp:224-309.
  - Sampler: argmax / multinomial / top-p with xorshift RNG
    — reference src/tokenizer.cpp:392-512.
  - ChatTemplateGenerator llama2/llama3/deepSeek3/chatml with auto-detection
    — reference src/tokenizer.cpp:549-637.
  - EosDetector with MAYBE_EOS buffering — reference src/tokenizer.cpp:639-724.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass

import numpy as np

TOKENIZER_MAGIC = 0x567124

# header keys (reference src/tokenizer.hpp:21-33)
TOK_VERSION = 0
TOK_VOCAB_SIZE = 1
MAX_TOKEN_LENGTH = 2
BOS_ID = 3
EOS_ID = 4
PAD_ID = 5
CHAT_EOS_ID = 6
CHAT_TEMPLATE = 7
CHAT_STOP = 8
N_EOS_TOKENS = 9
ADD_BOS = 10


class Tokenizer:
    def __init__(self, path: str):
        with open(path, "rb") as f:
            magic = struct.unpack("<i", f.read(4))[0]
            if magic != TOKENIZER_MAGIC:
                raise ValueError(f"invalid tokenizer magic 0x{magic:X}")
            header_size = struct.unpack("<i", f.read(4))[0]
            n_kv = (header_size - 8) // 4
            kv = struct.unpack(f"<{n_kv}i", f.read(n_kv * 4))

            self.vocab_size = 0
            self.max_token_length = 0
            self.bos_id = -1
            self.add_bos = True
            self.eos_token_ids: list[int] = []
            self.chat_template: str | None = None
            version = -1
            chat_template_len = -1
            n_eos = 0
            i = 0
            while i < n_kv:
                key, value = kv[i], kv[i + 1]
                i += 2
                if key == TOK_VERSION:
                    version = value
                elif key == TOK_VOCAB_SIZE:
                    self.vocab_size = value
                elif key == MAX_TOKEN_LENGTH:
                    self.max_token_length = value
                elif key == BOS_ID:
                    self.bos_id = value
                elif key in (EOS_ID, CHAT_EOS_ID):
                    self.eos_token_ids.append(value)
                elif key == CHAT_TEMPLATE:
                    chat_template_len = value
                elif key == CHAT_STOP:
                    f.seek(value, 1)
                elif key == PAD_ID:
                    pass
                elif key == N_EOS_TOKENS:
                    n_eos = value
                elif key == ADD_BOS:
                    self.add_bos = value == 1
                else:
                    raise ValueError(f"invalid tokenizer header key {key}")
            if version != 1:
                raise ValueError("unsupported tokenizer version (regenerate the .t file)")

            if chat_template_len > 0:
                self.chat_template = f.read(chat_template_len).decode("utf-8", "replace")
            for _ in range(n_eos):
                self.eos_token_ids.append(struct.unpack("<i", f.read(4))[0])

            self.vocab: list[bytes] = []
            self.scores = np.empty(self.vocab_size, dtype=np.float32)
            for t in range(self.vocab_size):
                score, length = struct.unpack("<fI", f.read(8))
                self.scores[t] = score
                self.vocab.append(f.read(length))

        # regular vs special split (reference src/tokenizer.cpp:139-156)
        self.regular_vocab_size = self.bos_id if self.bos_id >= 0 else self.vocab_size
        self._regular_index: dict[bytes, int] = {}
        for t in range(self.regular_vocab_size - 1, -1, -1):
            self._regular_index[self.vocab[t]] = t
        self._special = [(self.vocab[t], t)
                         for t in range(self.regular_vocab_size, self.vocab_size)]
        self._decode_buf = bytearray()
        # native C++ encoder from the extension (reference tokenizer is C++);
        # fall back to the pure-Python path when the extension isn't built
        self._native = None
        try:
            from .ops import hip_ops
            k = hip_ops()
            self._native = k.BpeEncoder(list(self.vocab),
                                        [float(s) for s in self.scores],
                                        self.regular_vocab_size)
        except Exception:  # noqa: BLE001
            self._native = None

    # ---------------------------------------------------------- encode

    def encode(self, text: str | bytes, is_start: bool = True,
               add_special_tokens: bool = True) -> list[int]:
        data = text.encode("utf-8") if isinstance(text, str) else bytes(text)
        tokens: list[int] = []
        if is_start and self.add_bos and self.bos_id >= 0:
            tokens.append(self.bos_id)

        if self._native is not None:
            tokens.extend(self._native.encode(data, add_special_tokens))
            return tokens

        buf = bytearray()
        i = 0
        n = len(data)
        while i < n:
            if add_special_tokens:
                sp = self._find_special(data, i)
                if sp is not None:
                    assert not buf, "unencodable byte run before special token"
                    tokens.append(sp)
                    i += len(self.vocab[sp])
                    continue
            buf.append(data[i])
            i += 1
            tid = self._regular_index.get(bytes(buf))
            if tid is not None:
                tokens.append(tid)
                buf.clear()
        if buf:
            raise ValueError(f"cannot encode byte run {bytes(buf)!r}")

        # greedy merge: repeatedly merge the adjacent pair whose concatenation
        # is the vocab token with the best score (tokenizer.cpp:352-379)
        while True:
            best_score = -1e10
            best_id = -1
            best_idx = -1
            for j in range(len(tokens) - 1):
                merged = self.vocab[tokens[j]] + self.vocab[tokens[j + 1]]
                tid = self._regular_index.get(merged)
                if tid is not None and self.scores[tid] > best_score:
                    best_score = float(self.scores[tid])
                    best_id = tid
                    best_idx = j
            if best_idx == -1:
                break
            tokens[best_idx: best_idx + 2] = [best_id]
        return tokens

    def _find_special(self, data: bytes, pos: int) -> int | None:
        for piece, tid in self._special:
            if data.startswith(piece, pos):
                return tid
        return None

    # ---------------------------------------------------------- decode

    def is_eos(self, token: int) -> bool:
        return token in self.eos_token_ids

    def reset_decoder(self) -> None:
        self._decode_buf.clear()

    def decode(self, token: int) -> str | None:
        """Streaming decode of one token; returns printable text or None
        while mid-UTF-8-sequence (reference src/tokenizer.cpp:291-309)."""
        if token == self.bos_id or token >= self.vocab_size:
            return None
        if self.is_eos(token):
            if self._decode_buf:
                out = self._decode_buf.decode("utf-8", "replace")
                self._decode_buf.clear()
                return out
            return None
        self._decode_buf.extend(self.vocab[token])
        # emit up to the last complete UTF-8 sequence; keep the tail buffered
        cut = len(self._decode_buf)
        for back in range(1, min(4, cut) + 1):
            b = self._decode_buf[cut - back]
            if b < 0x80:
                break  # ASCII tail is complete
            if b >= 0xC0:  # lead byte of a multi-byte sequence
                need = 2 if b < 0xE0 else 3 if b < 0xF0 else 4
                if back < need:
                    cut -= back  # incomplete sequence: hold it back
                break
        if cut == 0:
            return None
        out = self._decode_buf[:cut].decode("utf-8", "replace")
        del self._decode_buf[:cut]
        return out if out else None

    def piece(self, token: int) -> bytes:
        return self.vocab[token] if 0 <= token < self.vocab_size else b""


def write_tokenizer(path: str, vocab: list[bytes], scores, bos_id: int,
                    add_bos: bool, eos_tokens: list[int],
                    chat_template: str | None = None) -> None:
    """Serialize a .t file (converter/tokenizer-writer.py:3-57 semantics)."""
    tmpl = chat_template.encode("utf-8") if chat_template else None
    kv = [
        (BOS_ID, bos_id),
        (TOK_VERSION, 1),
        (TOK_VOCAB_SIZE, len(vocab)),
        (MAX_TOKEN_LENGTH, max(len(t) for t in vocab)),
    ]
    if tmpl:
        kv.append((CHAT_TEMPLATE, len(tmpl)))
    kv.append((N_EOS_TOKENS, len(eos_tokens)))
    kv.append((ADD_BOS, 1 if add_bos else 0))
    data = b"".join(struct.pack("<ii", k, v) for k, v in kv)
    with open(path, "wb") as f:
        f.write(struct.pack("<ii", TOKENIZER_MAGIC, 8 + len(data)))
        f.write(data)
        if tmpl:
            f.write(tmpl)
        for e in eos_tokens:
            f.write(struct.pack("<i", e))
        for piece, score in zip(vocab, scores):
            assert len(piece) > 0
            f.write(struct.pack("<fI", float(score), len(piece)))
            f.write(piece)


# ---------------------------------------------------------------- sampler

def _xorshift_u32(state: int) -> tuple[int, int]:
    """xorshift* RNG (reference src/tokenizer.cpp:25-31)."""
    state &= (1 << 64) - 1
    state ^= state >> 12
    state ^= (state << 25) & ((1 << 64) - 1)
    state ^= state >> 27
    return ((state * 0x2545F4914F6CDD1D) & ((1 << 64) - 1)) >> 32, state


class Sampler:
    """Temperature / top-p sampler (reference src/tokenizer.cpp:392-512)."""

    def __init__(self, vocab_size: int, temperature: float, topp: float, seed: int):
        self.vocab_size = vocab_size
        self.temperature = temperature
        self.topp = topp
        self.state = seed if seed else 1

    def set_seed(self, seed: int) -> None:
        self.state = seed if seed else 1

    def set_temp(self, temperature: float) -> None:
        self.temperature = temperature

    def _random_f32(self) -> float:
        u, self.state = _xorshift_u32(self.state)
        return (u >> 8) / 16777216.0

    def sample_torch(self, logits) -> int:
        """Device-side sampling for GPU logits (temperature/softmax/top-p all
        on device; only the chosen token id crosses PCIe). Same xorshift coin
        as the CPU path."""
        import torch
        logits = logits.reshape(-1)[: self.vocab_size]
        if self.temperature == 0.0:
            return int(torch.argmax(logits).item())
        p = torch.softmax(logits.float() / self.temperature, dim=-1)
        coin = self._random_f32()
        if self.topp <= 0 or self.topp >= 1:
            cdf = torch.cumsum(p, dim=-1)
            return int(torch.searchsorted(cdf, torch.tensor(coin, device=p.device),
                                          right=True)
                       .clamp(0, self.vocab_size - 1).item())
        # mirror the numpy path exactly (cutoff pre-filter, stable sort,
        # right-bisect) so GPU and CPU ranks stay in sampling lockstep
        cutoff = (1.0 - self.topp) / (self.vocab_size - 1)
        idx = torch.nonzero(p >= cutoff).reshape(-1)
        order = idx[torch.argsort(-p[idx], stable=True)]
        probs = p[order]
        c = torch.cumsum(probs, dim=-1)
        last = int(torch.searchsorted(c, torch.tensor(self.topp, device=p.device),
                                      right=True).item())
        last = min(last, order.numel() - 1)
        r = coin * float(c[last].item())
        pick = int(torch.searchsorted(c[: last + 1],
                                      torch.tensor(r, device=p.device),
                                      right=True).clamp(0, last).item())
        return int(order[pick].item())

    def sample(self, logits) -> int:
        import torch
        if isinstance(logits, torch.Tensor) and logits.is_cuda:
            return self.sample_torch(logits)
        logits = np.asarray(logits, dtype=np.float32).reshape(-1)[: self.vocab_size]
        if self.temperature == 0.0:
            return int(np.argmax(logits))
        x = logits / self.temperature
        x = x - x.max()
        p = np.exp(x)
        p /= p.sum()
        coin = self._random_f32()
        if self.topp <= 0 or self.topp >= 1:
            cdf = np.cumsum(p)
            return int(np.searchsorted(cdf, coin, side="right").clip(0, self.vocab_size - 1))
        # top-p (nucleus): smallest prefix of the sorted probs exceeding topp
        cutoff = (1.0 - self.topp) / (self.vocab_size - 1)
        idx = np.nonzero(p >= cutoff)[0]
        order = idx[np.argsort(-p[idx], kind="stable")]
        probs = p[order]
        c = np.cumsum(probs)
        last = int(np.searchsorted(c, self.topp, side="right"))
        last = min(last, len(order) - 1)
        r = coin * c[last]
        pick = int(np.searchsorted(c[: last + 1], r, side="right").clip(0, last))
        return int(order[pick])


# ---------------------------------------------------------------- chat

TEMPLATE_UNKNOWN = 0
TEMPLATE_LLAMA2 = 1
TEMPLATE_LLAMA3 = 2
TEMPLATE_DEEP_SEEK3 = 3
TEMPLATE_CHATML = 4

_TEMPLATE_NAMES = {
    "llama2": TEMPLATE_LLAMA2,
    "llama3": TEMPLATE_LLAMA3,
    "deepSeek3": TEMPLATE_DEEP_SEEK3,
    "chatml": TEMPLATE_CHATML,
}


@dataclass
class ChatItem:
    role: str
    message: str


@dataclass
class GeneratedChat:
    content: str
    public_prompt: str | None = None


class ChatTemplateGenerator:
    """Chat formatting with auto-detection (reference src/tokenizer.cpp:549-637)."""

    def __init__(self, template_type: int, chat_template: str | None, eos: str):
        if template_type == TEMPLATE_UNKNOWN:
            if chat_template is None:
                raise ValueError("the tokenizer does not include a chat template")
            if "[INST]" in chat_template:
                template_type = TEMPLATE_LLAMA2
            elif "<|start_header_id|>" in chat_template:
                template_type = TEMPLATE_LLAMA3
            elif "<｜Assistant｜>" in chat_template:
                template_type = TEMPLATE_DEEP_SEEK3
            elif "<|im_start|>" in chat_template:
                template_type = TEMPLATE_CHATML
            else:
                raise ValueError("unsupported chat template")
        self.type = template_type
        self.eos = eos

    def generate(self, items: list[ChatItem],
                 append_generation_prompt: bool = True) -> GeneratedChat:
        out = []
        public_prompt = None
        if self.type == TEMPLATE_LLAMA2:
            i = 0
            if len(items) >= 2 and items[0].role == "system" and items[1].role == "user":
                out.append("[INST] <<SYS>>\n" + items[0].message + "\n<</SYS>>\n\n"
                           + items[1].message + " [/INST]" + self.eos)
                i = 2
            for it in items[i:]:
                if it.role == "assistant":
                    out.append(it.message + self.eos)
                elif it.role == "user":
                    out.append("[INST] " + it.message + " [/INST]" + self.eos)
        elif self.type == TEMPLATE_LLAMA3:
            for it in items:
                out.append("<|start_header_id|>" + it.role + "<|end_header_id|>\n\n"
                           + it.message + self.eos)
            if append_generation_prompt:
                out.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
        elif self.type == TEMPLATE_DEEP_SEEK3:
            i = 0
            if items and items[0].role == "system":
                out.append(items[0].message)
                i = 1
            for it in items[i:]:
                if it.role == "user":
                    out.append("<｜User｜>" + it.message)
                elif it.role == "assistant":
                    out.append("<｜Assistant｜>" + it.message)
            if append_generation_prompt:
                out.append("<｜Assistant｜><think>\n")
                public_prompt = "<think>\n"
        elif self.type == TEMPLATE_CHATML:
            for it in items:
                if it.role in ("system", "user", "assistant"):
                    out.append("<|im_start|>" + it.role + "\n" + it.message + "<|im_end|>\n")
                # quirk kept for parity: the reference appends the generation
                # prompt INSIDE the item loop (tokenizer.cpp:615-627), i.e.
                # once per message, not once at the end
                if append_generation_prompt:
                    out.append("<|im_start|>assistant\n")
        return GeneratedChat("".join(out), public_prompt)


def chat_stops(tokenizer: Tokenizer) -> list[str]:
    """EOS pieces usable as text stops (reference TokenizerChatStops)."""
    return [tokenizer.vocab[t].decode("utf-8", "replace")
            for t in tokenizer.eos_token_ids]


# ---------------------------------------------------------------- eos detector

MAYBE_EOS = 0
EOS = 1
NOT_EOS = 2


class EosDetector:
    """Streaming stop-sequence detector (reference src/tokenizer.cpp:639-724)."""

    def __init__(self, tokens: list[int], pieces: list[str],
                 padding_left: int = 0, padding_right: int = 0):
        self.tokens = list(tokens)
        self.pieces = [p.encode("utf-8") if isinstance(p, str) else p for p in pieces]
        self.padding_left = padding_left
        self.padding_right = padding_right
        self.buffer = bytearray()
        self.eos_pos = -1

    def is_eos(self, token_id: int) -> bool:
        return token_id in self.tokens

    def append(self, token_id: int, piece: str | bytes | None) -> int:
        if piece is not None:
            self.buffer.extend(piece.encode("utf-8") if isinstance(piece, str) else piece)
        if self.is_eos(token_id):
            self.eos_pos = len(self.buffer)
            return EOS
        self.eos_pos = -1
        blen = len(self.buffer)
        for p in self.pieces:
            psize = len(p)
            if blen > psize + self.padding_left + self.padding_right:
                continue
            for lo in range(self.padding_left + 1):
                n = blen - lo
                if n == 0 or n > psize + self.padding_right:
                    continue
                n = min(n, psize)
                if self.buffer[lo: lo + n] == p[:n]:
                    if n == psize:
                        self.eos_pos = lo
                        del self.buffer[lo:]
                        return EOS
                    return MAYBE_EOS
        return NOT_EOS

    def get_delta(self) -> str | None:
        if not self.buffer:
            return None
        if self.eos_pos == 0:
            return None
        return self.buffer.decode("utf-8", "replace")

    def reset(self) -> None:
        self.buffer.clear()
        self.eos_pos = -1
