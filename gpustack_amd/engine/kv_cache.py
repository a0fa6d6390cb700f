"""Paged KV-cache pool for 288 GB HBM3E.

Pool layout per layer: [num_blocks, num_kv_heads, block_size, head_dim] bf16
— chosen so the decode kernel's per-wave token stripes are contiguous 1 KiB
reads (see ops/csrc/attn_decode.hip). The allocator is a free-list over
block ids shared by all layers (block i of every layer belongs to the same
logical page, so one block table drives all layers).
"""
from __future__ import annotations

import contextlib

import torch

from .config import EngineConfig


def _nullctx():
    return contextlib.nullcontext()


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


class CachingBlockAllocator(BlockAllocator):
    """Ref-counted allocator with a prefix cache (vLLM-style automatic
    prefix caching, re-designed for this engine): blocks released by a
    sequence stay resident (LRU-evictable) while registered under their
    content hash; a later prompt sharing the prefix re-acquires them with
    a refcount instead of recomputing the KV. Shared blocks are read-only
    by construction: the block holding position n-1 of any prompt is never
    registered, so every KV write lands in an owned block."""

    def __init__(self, num_blocks: int):
        super().__init__(num_blocks)
        self.ref: dict[int, int] = {}
        self.by_hash: dict[int, int] = {}
        self.hash_of: dict[int, int] = {}
        from collections import OrderedDict

        self.lru: "OrderedDict[int, None]" = OrderedDict()
        self.hits = 0
        self.misses = 0

    @property
    def num_free(self) -> int:
        return len(self.free_list) + len(self.lru)

    def _evict_one(self) -> int:
        blk, _ = self.lru.popitem(last=False)
        h = self.hash_of.pop(blk, None)
        if h is not None:
            self.by_hash.pop(h, None)
        return blk

    def allocate(self, n: int) -> list[int]:
        out: list[int] = []
        for _ in range(n):
            if self.free_list:
                b = self.free_list.pop()
            elif self.lru:
                b = self._evict_one()
            else:
                self.free_list.extend(reversed(out))  # roll back
                for b2 in out:
                    self.ref.pop(b2, None)
                raise RuntimeError("out of KV blocks")
            self.ref[b] = 1
            out.append(b)
        return out

    def free(self, blocks: list[int]) -> None:
        for b in blocks:
            r = self.ref.get(b, 1) - 1
            if r > 0:
                self.ref[b] = r
                continue
            self.ref.pop(b, None)
            if b in self.hash_of:
                self.lru[b] = None        # evictable, cache entry kept
                self.lru.move_to_end(b)
            else:
                self.free_list.append(b)

    def acquire_cached(self, h: int) -> int | None:
        b = self.by_hash.get(h)
        if b is None:
            self.misses += 1
            return None
        if b in self.lru:                 # resurrect from evictable set
            del self.lru[b]
            self.ref[b] = 1
        else:
            self.ref[b] = self.ref.get(b, 0) + 1
        self.hits += 1
        return b

    def register(self, block: int, h: int) -> None:
        if h in self.by_hash or block in self.hash_of:
            return  # first writer wins; duplicates stay plain
        self.by_hash[h] = block
        self.hash_of[block] = h


def block_hashes(tokens, block_size: int) -> list[int]:
    """Content-chain hashes for every FULL block of `tokens`."""
    import zlib

    out: list[int] = []
    h = 0
    for i in range(len(tokens) // block_size):
        blk = tokens[i * block_size:(i + 1) * block_size]
        h = zlib.crc32(repr(blk).encode(), h)
        out.append(h)
    return out


class KVCache:
    def __init__(self, cfg: EngineConfig, num_blocks: int,
                 device: str | torch.device, num_layers: int | None = None,
                 host_blocks: int | None = None):
        spec = cfg.spec
        self.block_size = cfg.block_size
        self.num_blocks = num_blocks
        kv_heads = max(1, spec.num_kv_heads // cfg.tp_size)
        self.kv_heads = kv_heads
        self.head_dim = spec.head_dim
        # pipeline stages hold KV only for their own layers
        self.num_layers = num_layers if num_layers is not None else spec.num_layers
        # MLA (DeepSeek): ONE compressed latent row per token — pool is
        # [nblocks, 1, BS, kv_lora_rank + qk_rope] and there is no V pool
        # (v_caches stay as zero-size placeholders so the layer-forward
        # signature is uniform)
        self.mla = bool(spec.kv_lora_rank)
        if self.mla:
            kv_heads = 1
            self.kv_heads = 1
            lat = spec.kv_lora_rank + spec.qk_rope_head_dim
            shape = (num_blocks, 1, cfg.block_size, lat)
        else:
            shape = (num_blocks, kv_heads, cfg.block_size, spec.head_dim)
        if cfg.kv_cache_dtype == "fp8":
            dtype = torch.float8_e4m3fn
        else:
            dtype = getattr(torch, cfg.dtype)
        self.kv_dtype = dtype
        self.k_caches = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(self.num_layers)
        ]
        if self.mla:
            self.v_caches = [
                torch.zeros(0, dtype=dtype, device=device)
                for _ in range(self.num_layers)
            ]
        else:
            self.v_caches = [
                torch.zeros(shape, dtype=dtype, device=device) for _ in range(self.num_layers)
            ]
        # last block reserved as the graph-padding scratch block (see
        # engine/graph_runner.py): padded rows write/read there, never live KV
        self.pad_block = num_blocks - 1
        if getattr(cfg, "enable_prefix_caching", False):
            self.allocator = CachingBlockAllocator(max(1, num_blocks - 1))
        else:
            self.allocator = BlockAllocator(max(1, num_blocks - 1))

        # ---- host-DRAM offload tier (extended_kv_cache) -----------------
        # Pinned host pool; swap-out/in run as async copies on a side HIP
        # stream (reference semantics: LMCache/HiCache tiering,
        # schemas/models.py:204-215, re-done natively per SURVEY.md §5.7).
        self.is_cuda = torch.device(device).type == "cuda"
        if host_blocks is None:
            host_blocks = self.compute_host_blocks(cfg, self.num_layers,
                                                   kv_heads)
        if self.mla:
            host_blocks = 0  # MLA latent offload tier is an r3 item
        self.host_blocks = host_blocks
        if host_blocks > 0:
            self.host_pool = torch.zeros(
                (host_blocks, self.num_layers, 2, kv_heads, cfg.block_size,
                 spec.head_dim),
                dtype=self.kv_dtype, pin_memory=self.is_cuda,
            )
            self.host_allocator = BlockAllocator(host_blocks)
            self.side_stream = torch.cuda.Stream() if self.is_cuda else None
        else:
            self.host_pool = None
            self.host_allocator = None
            self.side_stream = None

    # ---- offload tier ----------------------------------------------------

    def can_swap_out(self, n: int) -> bool:
        return self.host_allocator is not None and self.host_allocator.num_free >= n

    def swap_out(self, gpu_blocks: list[int]) -> list[int]:
        """Copy blocks D2H on the side stream and free the GPU blocks."""
        hblocks = self.host_allocator.allocate(len(gpu_blocks))
        if self.side_stream is not None:
            # D2H must observe the compute stream's latest KV writes
            self.side_stream.wait_stream(torch.cuda.current_stream())
        ctx = torch.cuda.stream(self.side_stream) if self.side_stream else _nullctx()
        with ctx:
            for g, h in zip(gpu_blocks, hblocks):
                for li in range(self.num_layers):
                    self.host_pool[h, li, 0].copy_(self.k_caches[li][g], non_blocking=True)
                    self.host_pool[h, li, 1].copy_(self.v_caches[li][g], non_blocking=True)
        if self.side_stream is not None:
            self.side_stream.synchronize()  # blocks must land before reuse
        self.allocator.free(gpu_blocks)
        return hblocks

    def swap_in(self, host_blocks: list[int]) -> list[int]:
        """Allocate GPU blocks and copy H2D; frees the host blocks."""
        gpu_blocks = self.allocator.allocate(len(host_blocks))
        ctx = torch.cuda.stream(self.side_stream) if self.side_stream else _nullctx()
        with ctx:
            for h, g in zip(host_blocks, gpu_blocks):
                for li in range(self.num_layers):
                    self.k_caches[li][g].copy_(self.host_pool[h, li, 0], non_blocking=True)
                    self.v_caches[li][g].copy_(self.host_pool[h, li, 1], non_blocking=True)
        if self.side_stream is not None:
            # compute stream must observe the blocks before attention reads
            torch.cuda.current_stream().wait_stream(self.side_stream)
        self.host_allocator.free(host_blocks)
        return gpu_blocks

    @staticmethod
    def compute_host_blocks(cfg: EngineConfig, num_layers: int,
                            kv_heads: int | None = None) -> int:
        """Host-DRAM offload pool size (blocks) for `kv_offload_gb`.

        Exposed so the runner can take a cross-rank MIN before building
        the pool (PP stages have different layer counts)."""
        if cfg.kv_offload_gb <= 0:
            return 0
        spec = cfg.spec
        if kv_heads is None:
            kv_heads = max(1, spec.num_kv_heads // cfg.tp_size)
        per_block = (2 * num_layers * kv_heads * cfg.block_size
                     * spec.head_dim * 2)
        return int(cfg.kv_offload_gb * 2**30) // per_block

    @staticmethod
    def compute_num_blocks(cfg: EngineConfig, free_bytes: int,
                           num_layers: int | None = None) -> int:
        spec = cfg.spec
        nl = num_layers if num_layers is not None else spec.num_layers
        kv_heads = max(1, spec.num_kv_heads // cfg.tp_size)
        esize = 1 if cfg.kv_cache_dtype == "fp8" else 2
        if spec.kv_lora_rank:
            per_block = (nl * cfg.block_size
                         * (spec.kv_lora_rank + spec.qk_rope_head_dim)
                         * esize)
        else:
            per_block = (
                2 * nl * kv_heads * cfg.block_size * spec.head_dim * esize
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
