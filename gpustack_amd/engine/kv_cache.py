"""Paged KV-cache pool for 288 GB HBM3E.

Pool layout per layer: [num_blocks, num_kv_heads, block_size, head_dim] bf16
— chosen so the decode kernel's per-wave token stripes are contiguous 1 KiB
reads (see ops/csrc/attn_decode.hip). The allocator is a free-list over
block ids shared by all layers (block i of every layer belongs to the same
logical page, so one block table drives all layers).
"""
from __future__ import annotations

import torch

from .config import EngineConfig


class BlockAllocator:
    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self.free_list: list[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def num_free(self) -> int:
        return len(self.free_list)

    def allocate(self, n: int) -> list[int]:
        if n > len(self.free_list):
            raise RuntimeError("out of KV blocks")
        out = [self.free_list.pop() for _ in range(n)]
        return out

    def free(self, blocks: list[int]) -> None:
        self.free_list.extend(reversed(blocks))


class KVCache:
    def __init__(self, cfg: EngineConfig, num_blocks: int, device: str | torch.device):
        spec = cfg.spec
        self.block_size = cfg.block_size
        self.num_blocks = num_blocks
        kv_heads = max(1, spec.num_kv_heads // cfg.tp_size)
        shape = (num_blocks, kv_heads, cfg.block_size, spec.head_dim)
        dtype = getattr(torch, cfg.dtype)
        self.k_caches = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(spec.num_layers)
        ]
        self.v_caches = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(spec.num_layers)
        ]
        # last block reserved as the graph-padding scratch block (see
        # engine/graph_runner.py): padded rows write/read there, never live KV
        self.pad_block = num_blocks - 1
        self.allocator = BlockAllocator(max(1, num_blocks - 1))

    @staticmethod
    def compute_num_blocks(cfg: EngineConfig, free_bytes: int) -> int:
        spec = cfg.spec
        kv_heads = max(1, spec.num_kv_heads // cfg.tp_size)
        per_block = (
            2 * spec.num_layers * kv_heads * cfg.block_size * spec.head_dim * 2
        )
        return max(1, int(free_bytes * cfg.gpu_memory_utilization) // per_block)

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def slots_for(self, block_table: list[int], start_token: int, n: int) -> list[int]:
        """Global slot ids for token positions [start_token, start_token+n)."""
        bs = self.block_size
        out = []
        for t in range(start_token, start_token + n):
            out.append(block_table[t // bs] * bs + t % bs)
        return out
