"""Tensor-parallel communication over RCCL / xGMI.

One process per GPU; `torch.distributed` with backend "nccl" IS RCCL on
ROCm. On an MI355X node each GPU has 7 point-to-point xGMI links
(~153 GB/s each), so ring all-reduce is single-link-bound: the decode-step
all-reduce tensors are small (batch x hidden x 2 B), which RCCL handles
with its low-latency (LL) protocol — no custom one-shot kernel needed at
TP<=8 for v1 (measured before optimizing, per the guide's methodology).

The reference never runs collectives itself (NCCL lives
inside vLLM); here they are first-party.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist


class Communicator:
    """TP group communicator; degenerates to no-ops at world size 1."""

    def __init__(self, tp_size: int = 1, tp_rank: int = 0, group=None):
        self.tp_size = tp_size
        self.tp_rank = tp_rank
        self.group = group

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.tp_size > 1:
            dist.all_reduce(t, group=self.group)
        return t

    def all_gather(self, t: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if self.tp_size == 1:
            return t
        parts = [torch.empty_like(t) for _ in range(self.tp_size)]
        dist.all_gather(parts, t, group=self.group)
        return torch.cat(parts, dim=dim)

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.tp_size > 1:
            dist.broadcast(t, src=src, group=self.group)
        return t


_COMM = Communicator()


def get_communicator() -> Communicator:
    return _COMM


def init_tp(tp_size: int, tp_rank: int, master_port: int | None = None,
            backend: str | None = None, device_id: int | None = None,
            master_addr: str | None = None) -> Communicator:
    """Initialize the TP process group (rank bootstrap via TCP store on
    127.0.0.1, replacing the master-port scheme the reference's port
    allocator models — serve_manager.py:1685-1737)."""
    global _COMM
    if tp_size <= 1:
        _COMM = Communicator()
        return _COMM
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        if master_addr:
            os.environ["MASTER_ADDR"] = master_addr
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        if master_port is not None:
            os.environ["MASTER_PORT"] = str(master_port)
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend, world_size=tp_size, rank=tp_rank)
    if backend == "nccl" and device_id is not None:
        torch.cuda.set_device(device_id)
    _COMM = Communicator(tp_size, tp_rank)
    return _COMM
