"""Tensor-parallel collectives.

This replaces the reference's TCP socket mesh + hand-rolled sync loops
(nn-network.cpp:541-632) with torch.distributed collectives — RCCL over
xGMI on GPUs (backend "nccl" IS RCCL on ROCm), gloo on CPU.

The Q80-quantized sync: each rank packs its full-dim partial output into
the Q80 wire layout on-device, the packed slices are all-gathered, and a
merge-add dequantizes + sums all slices (reference SYNC_NODE_SLICES +
OP_MERGE_ADD, nn-network.cpp:568-600 / nn-cpu-ops.cpp:920-957).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


class Comm:
    rank: int = 0
    world: int = 1

    def allreduce_(self, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def all_gather(self, out: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
        """Gather x from every rank into out[world, *x.shape]."""
        raise NotImplementedError

    def broadcast_(self, x: torch.Tensor, src: int = 0) -> torch.Tensor:
        raise NotImplementedError

    def barrier(self) -> None:
        pass


class SingleComm(Comm):
    """world=1: every collective is a no-op."""

    def allreduce_(self, x):
        return x

    def all_gather(self, out, x):
        out.copy_(x.unsqueeze(0))
        return out

    def broadcast_(self, x, src=0):
        return x


class DistComm(Comm):
    def __init__(self, group=None):
        assert dist.is_initialized()
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)

    def allreduce_(self, x):
        dist.all_reduce(x, op=dist.ReduceOp.SUM, group=self.group)
        return x

    def all_gather(self, out, x):
        dist.all_gather_into_tensor(out.view(-1), x.reshape(-1), group=self.group)
        return out

    def broadcast_(self, x, src=0):
        dist.broadcast(x, src=src, group=self.group)
        return x

    def barrier(self):
        dist.barrier(group=self.group)


def init_dist_comm(backend: str | None = None, timeout_s: int = 300) -> Comm:
    """Initialize torch.distributed from torchrun env vars and return a Comm.

    One process per GPU; nccl(=RCCL) when GPUs are visible, gloo otherwise.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world == 1:
        return SingleComm()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        # bind the device BEFORE init so the first collective (and barrier's
        # device guess) never lands every rank on GPU 0
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    if not dist.is_initialized():
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=timeout_s))
    return DistComm()
