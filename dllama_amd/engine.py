"""Inference engine: prefill/decode orchestration, sampling, timing stats.

Mirrors the reference app loop (src/dllama.cpp:13-116): prompt evaluated in
batches of <= n_batches tokens through the same graph, then single-token
decode; reports eval/pred tokens-per-second the same way
(dllama.cpp:104-115).

Under TP every rank runs the engine in lockstep; logits are gathered on all
ranks and sampling is deterministic (same seed), so no extra token
broadcast is needed (replaces the reference LlmControlPacket root->worker
broadcast, app.cpp:197-230).
"""

from __future__ import annotations

import time
from dataclasses import dataclass

import torch

from .tokenizer import Sampler


@dataclass
class GenStats:
    prefill_tokens: int = 0
    prefill_time: float = 0.0
    decode_tokens: int = 0
    decode_time: float = 0.0

    @property
    def eval_tok_s(self) -> float:
        return self.prefill_tokens / self.prefill_time if self.prefill_time else 0.0

    @property
    def pred_tok_s(self) -> float:
        return self.decode_tokens / self.decode_time if self.decode_time else 0.0


class InferenceEngine:
    def __init__(self, model, tokenizer=None, sampler: Sampler | None = None,
                 n_batches: int = 32):
        self.model = model
        self.tokenizer = tokenizer
        self.sampler = sampler
        self.n_batches = n_batches
        self.pos = 0

    def reset(self, pos: int = 0) -> None:
        self.pos = pos

    def _sync_device(self):
        if getattr(self.model, "device", None) is not None:
            d = self.model.device
            if d.type == "cuda":
                torch.cuda.synchronize(d)

    def prefill(self, tokens: list[int], start_pos: int | None = None) -> torch.Tensor:
        """Feed prompt tokens in chunks of n_batches; returns logits of the
        last token [vocab]."""
        if start_pos is not None:
            self.pos = start_pos
        logits = None
        starts = list(range(0, len(tokens), self.n_batches))
        for i in starts:
            chunk = tokens[i: i + self.n_batches]
            t = torch.tensor(chunk, dtype=torch.int64)
            p = torch.arange(self.pos, self.pos + len(chunk), dtype=torch.int64)
            # intermediate prefill chunks don't need logits (the reference
            # computes them anyway; real serving shouldn't)
            self.model.skip_logits = i != starts[-1]
            try:
                logits = self.model.forward(t, p)
            finally:
                self.model.skip_logits = False
            self.pos += len(chunk)
        return logits[-1]

    def decode_one(self, token: int) -> torch.Tensor:
        """One decode step; returns logits [vocab] for the next token."""
        t = torch.tensor([token], dtype=torch.int64)
        p = torch.tensor([self.pos], dtype=torch.int64)
        logits = self.model.forward(t, p)
        self.pos += 1
        return logits[0]

    def generate(self, prompt_tokens: list[int], max_tokens: int,
                 on_token=None, stop_check=None) -> tuple[list[int], GenStats]:
        """Greedy/sampled generation. on_token(token_id) is called per new
        token; stop_check(token_id) -> bool ends generation."""
        stats = GenStats()
        assert len(prompt_tokens) >= 1

        self._sync_device()
        t0 = time.perf_counter()
        logits = self.prefill(prompt_tokens)
        self._sync_device()
        stats.prefill_time = time.perf_counter() - t0
        stats.prefill_tokens = len(prompt_tokens)

        out: list[int] = []
        sampler = self.sampler or Sampler(logits.shape[-1], 0.0, 0.9, 12345)
        t0 = time.perf_counter()
        token = sampler.sample(logits)
        out.append(token)
        if on_token:
            on_token(token)
        seq_len = getattr(getattr(self.model, "cfg", None), "seq_len", None)
        for _ in range(max_tokens - 1):
            if stop_check and stop_check(token):
                break
            if seq_len is not None and self.pos >= seq_len:
                break  # context window exhausted (reference clamps via seqLen)
            logits = self.decode_one(token)
            token = sampler.sample(logits)
            out.append(token)
            if on_token:
                on_token(token)
        self._sync_device()
        stats.decode_time = time.perf_counter() - t0
        stats.decode_tokens = len(out)
        return out, stats


def _to_numpy(x: torch.Tensor):
    return x.detach().float().cpu().numpy()
