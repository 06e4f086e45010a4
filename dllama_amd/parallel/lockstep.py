"""Root/follower lockstep execution for TP serving.

The API server cannot run one HTTP server per rank (every rank would bind
the port and sample independently). Instead rank 0 owns the request loop
and, like the reference root, broadcasts a control packet before every
forward so follower ranks replay it blindly; batch==0 is the stop signal
(reference LlmControlPacket root->worker broadcast and worker poll loop,
src/app.cpp:197-230).

The packet is (batch, position, skip_logits) + the token ids — everything a
rank needs to run the identical collective schedule. Sampling, detectors
and HTTP happen only on rank 0; followers never inspect logits.
"""

from __future__ import annotations

import torch

from .comm import Comm


def _ctrl_device(model):
    dev = getattr(model, "device", None)
    return dev if dev is not None and dev.type == "cuda" else torch.device("cpu")


class RootModel:
    """Rank-0 model wrapper: broadcast the control packet + tokens, then
    run the real forward. Drop-in for InferenceEngine's model."""

    def __init__(self, model, comm: Comm):
        self._model = model
        self._comm = comm
        self._dev = _ctrl_device(model)

    def forward(self, tokens: torch.Tensor, positions: torch.Tensor):
        m = self._model
        B = tokens.shape[0]
        ctrl = torch.tensor([B, int(positions[0]), int(bool(m.skip_logits))],
                            dtype=torch.int64, device=self._dev)
        self._comm.broadcast_(ctrl, src=0)
        tok = tokens.to(device=self._dev, dtype=torch.int64)
        self._comm.broadcast_(tok, src=0)
        return m.forward(tokens, positions)

    def stop_followers(self) -> None:
        """Release ranks > 0 from their follow loop (batch == 0 packet)."""
        ctrl = torch.zeros(3, dtype=torch.int64, device=self._dev)
        self._comm.broadcast_(ctrl, src=0)

    def __getattr__(self, name):
        return getattr(self._model, name)

    @property
    def skip_logits(self):
        return self._model.skip_logits

    @skip_logits.setter
    def skip_logits(self, v):
        self._model.skip_logits = v


def follower_loop(model, comm: Comm) -> int:
    """Ranks > 0: replay control packets until batch == 0 (reference worker
    serve loop, app.cpp:217-230). Returns the number of forwards run."""
    dev = _ctrl_device(model)
    n = 0
    while True:
        ctrl = torch.zeros(3, dtype=torch.int64, device=dev)
        comm.broadcast_(ctrl, src=0)
        B, pos, skip = int(ctrl[0]), int(ctrl[1]), int(ctrl[2])
        if B == 0:
            return n
        tok = torch.zeros(B, dtype=torch.int64, device=dev)
        comm.broadcast_(tok, src=0)
        model.skip_logits = bool(skip)
        try:
            model.forward(tok.cpu() if dev.type == "cpu" else tok,
                          torch.arange(pos, pos + B))
        finally:
            model.skip_logits = False
        n += 1
