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
    """TP (+PP, +CP) communicator; degenerates to no-ops at world size 1.

    Rank layout: global = (pp_rank * cp_size + cp_rank) * tp_size + tp_rank
    — TP groups stay contiguous so their RCCL rings ride node-local xGMI
    links, while PP boundaries (one point-to-point hidden-state transfer
    per stage per step) are the only traffic that crosses nodes. `group`
    is the TP subgroup, `cp_group` the prefill-context-parallel subgroup
    (ranks with the same tp index); PP send/recv and world broadcasts use
    the default group. CP (reference: vLLM --prefill-context-parallel-size,
    SURVEY.md §2.10) is mutually exclusive with PP for now."""

    def __init__(self, tp_size: int = 1, tp_rank: int = 0, group=None,
                 pp_size: int = 1, pp_rank: int = 0,
                 cp_size: int = 1, cp_rank: int = 0, cp_group=None):
        self.tp_size = tp_size
        self.tp_rank = tp_rank
        self.group = group
        self.pp_size = pp_size
        self.pp_rank = pp_rank
        self.cp_size = cp_size
        self.cp_rank = cp_rank
        self.cp_group = cp_group

    @property
    def world_size(self) -> int:
        return self.tp_size * self.pp_size * self.cp_size

    @property
    def world_rank(self) -> int:
        return (self.pp_rank * self.cp_size + self.cp_rank) * self.tp_size \
            + self.tp_rank

    @property
    def is_first_stage(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last_stage(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    @property
    def last_stage_rank(self) -> int:
        """Global rank of (last stage, tp_rank 0) — the sampling rank."""
        return (self.pp_size - 1) * self.cp_size * self.tp_size

    @property
    def cp_tail_rank(self) -> int:
        """Global rank of (cp_rank cp_size-1, tp_rank 0) — the rank that
        owns every sequence's tail chunk under CP prefill (floor-bound
        partition: chunk cp-1 = [(cp-1)*L//cp, L) is never empty)."""
        return (self.cp_size - 1) * self.tp_size

    def cp_all_gather_rows(self, t: torch.Tensor) -> torch.Tensor:
        """All-gather equal-shaped row blocks across the CP group and
        concatenate along dim 0 (rank order = ascending chunk position)."""
        if self.cp_size == 1:
            return t
        t = t.contiguous()
        parts = [torch.empty_like(t) for _ in range(self.cp_size)]
        dist.all_gather(parts, t, group=self.cp_group)
        return torch.cat(parts, dim=0)

    # -- pipeline point-to-point -------------------------------------------
    def send_hidden(self, t: torch.Tensor) -> None:
        dst = (self.pp_rank + 1) * self.tp_size + self.tp_rank
        dist.send(t.contiguous(), dst=dst)

    def recv_hidden(self, shape, dtype, device) -> torch.Tensor:
        src = (self.pp_rank - 1) * self.tp_size + self.tp_rank
        t = torch.empty(shape, dtype=dtype, device=device)
        dist.recv(t, src=src)
        return t

    def broadcast_world(self, t: torch.Tensor, src: int) -> torch.Tensor:
        if self.world_size > 1:
            dist.broadcast(t, src=src)
        return t

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
    """Initialize a pure-TP process group (rank bootstrap via TCP store on
    127.0.0.1, replacing the master-port scheme the reference's port
    allocator models — serve_manager.py:1685-1737)."""
    return init_parallel(tp_size, 1, tp_rank, master_port=master_port,
                         backend=backend, device_id=device_id,
                         master_addr=master_addr)


def init_parallel(tp_size: int, pp_size: int, global_rank: int,
                  master_port: int | None = None, backend: str | None = None,
                  device_id: int | None = None,
                  master_addr: str | None = None,
                  cp_size: int = 1) -> Communicator:
    """Initialize a TP x PP x CP process group
    (global = (pp*cp_size + cp) * tp_size + tp)."""
    global _COMM
    if cp_size > 1 and pp_size > 1:
        raise ValueError("prefill context parallelism (cp) cannot combine "
                         "with pipeline parallelism yet")
    world = tp_size * pp_size * cp_size
    if world <= 1:
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
        dist.init_process_group(backend=backend, world_size=world,
                                rank=global_rank)
    if backend == "nccl" and device_id is not None:
        torch.cuda.set_device(device_id)
    outer, tp_rank = divmod(global_rank, tp_size)
    pp_rank, cp_rank = divmod(outer, cp_size)
    tp_group = None
    if tp_size > 1 and (pp_size > 1 or cp_size > 1):
        # every rank must create every subgroup (collective contract)
        for o in range(pp_size * cp_size):
            g = dist.new_group(list(range(o * tp_size, (o + 1) * tp_size)))
            if o == outer:
                tp_group = g
    cp_group = None
    if cp_size > 1 and tp_size > 1:
        for t in range(tp_size):
            g = dist.new_group([c * tp_size + t for c in range(cp_size)])
            if t == tp_rank:
                cp_group = g
    _COMM = Communicator(tp_size, tp_rank, tp_group, pp_size, pp_rank,
                         cp_size, cp_rank, cp_group)
    return _COMM
