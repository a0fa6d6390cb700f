"""hipGraph capture of the decode step.

The profile (profiles/r01) showed the decode path dominated by host gaps
and per-launch overhead (~300 dispatches/step across 32 layers, GPU busy
~14%). Instead of a tracing compiler, the whole decode forward is captured
once per batch-size bucket into a hipGraph (torch.cuda.CUDAGraph == hipGraph
on ROCm) with static buffers; each step writes the batch into the buffers
and replays one graph.

Padding safety: rows beyond the live batch point at a reserved pad block
(last block of the pool) so the captured reshape_and_cache writes land in
scratch, never in live KV.
"""
from __future__ import annotations

import os

import torch

from ..models.llama import ForwardMeta
from .scheduler import ScheduledBatch

BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 160, 192, 224, 256,
           320, 384, 448, 512, 640, 768, 896, 1024]


class DecodeGraphRunner:
    def __init__(self, runner, want_hidden: bool = False):
        # want_hidden: also capture the post-final-norm hidden states —
        # the EAGLE verify forward needs them to condition the next draft
        # window (runner.execute's return_both path)
        self.want_hidden = want_hidden
        self.runner = runner
        cfg = runner.cfg
        dev = runner.device
        spec = getattr(cfg, "speculative", None)
        rps = 1 + int(spec.get("num_draft_tokens", 3)) if spec else 1
        self.max_bs = min(cfg.max_num_seqs * rps, BUCKETS[-1])
        self.buckets = [b for b in BUCKETS if b <= self.max_bs]
        if self.buckets[-1] != self.max_bs:
            self.buckets.append(self.max_bs)
        self.max_blocks = (cfg.max_model_len + cfg.block_size - 1) // cfg.block_size
        kv = runner.kv
        self.pad_block = kv.pad_block
        pad_slot = self.pad_block * cfg.block_size

        mb = self.max_bs
        self.tokens = torch.zeros(mb, dtype=torch.long, device=dev)
        self.positions = torch.zeros(mb, dtype=torch.long, device=dev)
        self.slots = torch.full((mb,), pad_slot, dtype=torch.long, device=dev)
        self.block_tables = torch.full(
            (mb, self.max_blocks), self.pad_block, dtype=torch.int32, device=dev
        )
        self.seq_lens = torch.ones(mb, dtype=torch.int32, device=dev)
        # pinned host staging
        pin = dev.type == "cuda"
        self.h_tokens = torch.zeros(mb, dtype=torch.long, pin_memory=pin)
        self.h_positions = torch.zeros(mb, dtype=torch.long, pin_memory=pin)
        self.h_slots = torch.zeros(mb, dtype=torch.long, pin_memory=pin)
        self.h_bt = torch.zeros(mb, self.max_blocks, dtype=torch.int32, pin_memory=pin)
        self.h_seq_lens = torch.ones(mb, dtype=torch.int32, pin_memory=pin)
        # cached numpy views of the pinned staging (shared memory)
        self.n_tokens = self.h_tokens.numpy()
        self.n_positions = self.h_positions.numpy()
        self.n_slots = self.h_slots.numpy()
        self.n_seq_lens = self.h_seq_lens.numpy()
        self._pad_slot = pad_slot
        self._prev_bs = 0
        # per-row cache for incremental block-table staging: (request_id,
        # nblocks) — decode batches are stable between steps, so most rows
        # need no host work at all
        self._row_state: list[tuple[str, int] | None] = [None] * mb
        # persistent — every tensor a captured graph references must outlive
        # the graph (an ephemeral arange here caused replay-time faults)
        self.logits_idx = torch.arange(mb, dtype=torch.long, device=dev)
        self.graphs: dict[int, torch.cuda.CUDAGraph] = {}
        self.outs: dict[int, torch.Tensor] = {}
        self.metas: dict[int, ForwardMeta] = {}

    def _meta(self, bs: int) -> ForwardMeta:
        return ForwardMeta(
            is_prefill=False,
            positions=self.positions[:bs],
            slot_mapping=self.slots[:bs],
            logits_indices=self.logits_idx[:bs],
            block_tables=self.block_tables[:bs],
            seq_lens=self.seq_lens[:bs],
        )

    def capture(self) -> None:
        model, kv = self.runner.model, self.runner.kv
        pool = torch.cuda.graph_pool_handle()
        for bs in reversed(self.buckets):  # largest first reserves the pool
            meta = self._meta(bs)
            model(self.tokens[:bs], meta, kv,
                  return_both=self.want_hidden)  # eager warmup (blas workspaces)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=pool):
                out = model(self.tokens[:bs], meta, kv,
                            return_both=self.want_hidden)
            self.graphs[bs] = g
            self.outs[bs] = out
            self.metas[bs] = meta  # keep every referenced tensor alive
        torch.cuda.synchronize()

    def can_run(self, batch: ScheduledBatch) -> bool:
        return ((not batch.is_prefill)
                and not batch.is_suffix
                and len(batch.seqs) * batch.rows_per_seq <= self.max_bs
                and bool(self.graphs))

    def run(self, batch: ScheduledBatch, token_src: torch.Tensor | None = None) -> torch.Tensor:
        import numpy as np

        rps = batch.rows_per_seq
        bs = len(batch.seqs) * rps
        bucket = next(b for b in self.buckets if b >= bs)
        # host staging (numpy views over pinned memory)
        if token_src is None:
            self.n_tokens[:bs] = batch.token_ids
        self.n_positions[:bs] = batch.positions
        self.n_slots[:bs] = batch.slot_mapping
        self.n_seq_lens[:bs] = np.asarray(batch.seq_lens, dtype=np.int32)
        maxb = 0
        dirty_lo, dirty_hi = self.max_bs, -1
        for si, s in enumerate(batch.seqs):
            nb = len(s.block_table)
            maxb = max(maxb, nb)
            state = (s.request_id, nb)
            for j in range(rps):
                i = si * rps + j
                if self._row_state[i] != state:
                    self.h_bt[i, :nb] = torch.tensor(s.block_table, dtype=torch.int32)
                    self._row_state[i] = state
                    dirty_lo = min(dirty_lo, i)
                    dirty_hi = max(dirty_hi, i)
        # pad rows dirtied by a previous (larger) batch
        hi = max(self._prev_bs, bucket)
        if hi > bs:
            self.n_tokens[bs:hi] = 0
            self.n_positions[bs:hi] = 0
            self.n_slots[bs:hi] = self._pad_slot
            self.n_seq_lens[bs:hi] = 1
            self.h_bt[bs:hi, 0] = self.pad_block
            for i in range(bs, hi):
                self._row_state[i] = None
            dirty_lo = min(dirty_lo, bs)
            dirty_hi = max(dirty_hi, hi - 1)
        self._prev_bs = bucket
        n = hi
        if token_src is not None:
            # async pipeline: previous step's sampled ids feed this step
            # device-to-device (same stream => ordered after the sampler)
            self.tokens[:bs].copy_(token_src[:bs])
            if n > bs:
                self.tokens[bs:n].copy_(self.h_tokens[bs:n], non_blocking=True)
        else:
            self.tokens[:n].copy_(self.h_tokens[:n], non_blocking=True)
        self.positions[:n].copy_(self.h_positions[:n], non_blocking=True)
        self.slots[:n].copy_(self.h_slots[:n], non_blocking=True)
        self.seq_lens[:n].copy_(self.h_seq_lens[:n], non_blocking=True)
        if dirty_hi >= 0:
            self.block_tables[dirty_lo:dirty_hi + 1, :max(maxb, 1)].copy_(
                self.h_bt[dirty_lo:dirty_hi + 1, :max(maxb, 1)], non_blocking=True
            )
        self.graphs[bucket].replay()
        if self.want_hidden:
            logits, hidden = self.outs[bucket]
            return logits[:bs], hidden[:bs]
        return self.outs[bucket][:bs]


def graphs_enabled() -> bool:
    return os.environ.get("GPUSTACK_AMD_NO_GRAPHS", "0") != "1"
