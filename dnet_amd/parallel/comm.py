"""torch.distributed bootstrap + ring point-to-point helpers.

MI355X-native replacement for the reference's gRPC StreamActivations ring
(reference: src/dnet/shard/adapters/ring.py): on ROCm the "nccl" backend IS
RCCL, and ring hops become RCCL send/recv over xGMI (7 p2p links x ~153 GB/s
per GPU; activation hops are tiny so latency dominates — persistent
communicator, pre-posted receives). On CPU test rigs the same code runs on
gloo with world_size>1.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_from_env() -> tuple[int, int, torch.device]:
    """Initialise torch.distributed from torchrun-style env vars.

    Returns (rank, world_size, device). Single-process (no env / world 1)
    returns without creating a process group.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world == 1:
        dev = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        if dev.type == "cuda":
            torch.cuda.set_device(dev)
        return 0, 1, dev
    use_gpu = torch.cuda.is_available()
    backend = "nccl" if use_gpu else "gloo"
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if use_gpu:
        torch.cuda.set_device(local_rank)
        dev = torch.device(f"cuda:{local_rank}")
    else:
        dev = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=600))
    return rank, world, dev


def barrier():
    if dist.is_initialized():
        dist.barrier()


class Ring:
    """Closed ring over ranks 0..world-1 (next = (r+1) % world)."""

    def __init__(self, rank: int, world: int, device: torch.device):
        self.rank = rank
        self.world = world
        self.device = device
        self.next = (rank + 1) % world
        self.prev = (rank - 1) % world

    def send(self, t: torch.Tensor, dst: int | None = None):
        dist.send(t.contiguous(), self.next if dst is None else dst)

    def recv(self, t: torch.Tensor, src: int | None = None):
        dist.recv(t, self.prev if src is None else src)

    def isend(self, t: torch.Tensor, dst: int | None = None):
        return dist.isend(t.contiguous(), self.next if dst is None else dst)

    def irecv(self, t: torch.Tensor, src: int | None = None):
        return dist.irecv(t, self.prev if src is None else src)
