"""Continuous-batching scheduler for the inference engine.

vLLM-v0-style iteration scheduling redesigned for the MI355X memory budget:
each step is either one varlen PREFILL batch (prefill-priority keeps TTFT
low at fixed QPS — the BASELINE.json metric) or one DECODE batch over all
running sequences. 288 GB HBM3E means preemption is rare; when the pool
does run out we preempt-by-recompute (free blocks, re-prefill later).
"""
from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field

import numpy as np

from .config import EngineConfig
from .kv_cache import KVCache
from .sequence import Sequence, SeqStatus


@dataclass
class ScheduledBatch:
    is_prefill: bool
    seqs: list[Sequence] = field(default_factory=list)
    # flat token ids / positions / KV-write slots for the whole batch
    # (lists for prefill; int64 numpy arrays for decode — single-pass host
    # assembly keeps the per-step gap small)
    token_ids: "list[int] | np.ndarray" = field(default_factory=list)
    positions: "list[int] | np.ndarray" = field(default_factory=list)
    slot_mapping: "list[int] | np.ndarray" = field(default_factory=list)
    seq_lens: list[int] = field(default_factory=list)   # context length per seq

    @property
    def num_tokens(self) -> int:
        return len(self.token_ids)


class Scheduler:
    def __init__(self, cfg: EngineConfig, kv: KVCache):
        self.cfg = cfg
        self.kv = kv
        self.waiting: deque[Sequence] = deque()
        self.running: list[Sequence] = []

    # -- queue ops ---------------------------------------------------------
    def add(self, seq: Sequence) -> None:
        self.waiting.append(seq)

    def abort(self, request_id: str) -> bool:
        for i, s in enumerate(self.running):
            if s.request_id == request_id:
                self._release(s)
                s.finish("abort")
                del self.running[i]
                return True
        for i, s in enumerate(list(self.waiting)):
            if s.request_id == request_id:
                s.finish("abort")
                self.waiting.remove(s)
                return True
        return False

    @property
    def num_unfinished(self) -> int:
        return len(self.waiting) + len(self.running)

    def has_work(self) -> bool:
        return self.num_unfinished > 0

    # -- scheduling --------------------------------------------------------
    def schedule(self) -> ScheduledBatch | None:
        batch = self._schedule_prefill()
        if batch is not None:
            return batch
        return self._schedule_decode()

    def _schedule_prefill(self) -> ScheduledBatch | None:
        if not self.waiting:
            return None
        batch = ScheduledBatch(is_prefill=True)
        budget = self.cfg.max_prefill_tokens
        while self.waiting and len(self.running) + len(batch.seqs) < self.cfg.max_num_seqs:
            seq = self.waiting[0]
            n = seq.num_tokens  # prompt + any generated tokens (preempted seqs)
            if batch.seqs and batch.num_tokens + n > budget:
                break
            nblocks = self.kv.blocks_needed(n)
            if nblocks > self.kv.allocator.num_free:
                break
            self.waiting.popleft()
            seq.block_table = self.kv.allocator.allocate(nblocks)
            seq.num_cached_tokens = 0
            seq.status = SeqStatus.RUNNING
            tokens = seq.all_token_ids
            batch.seqs.append(seq)
            batch.token_ids.extend(tokens)
            batch.positions.extend(range(n))
            batch.slot_mapping.extend(self.kv.slots_for(seq.block_table, 0, n))
            batch.seq_lens.append(n)
        if not batch.seqs:
            return None
        return batch

    def _schedule_decode(self) -> ScheduledBatch | None:
        if not self.running:
            return None
        batch = ScheduledBatch(is_prefill=False)
        # Ensure every running seq has a slot for its next token; preempt from
        # the back (most recent) on pool exhaustion.
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            pos = seq.num_tokens - 1          # position of the token to feed
            if (pos + 1) > len(seq.block_table) * self.kv.block_size:
                try:
                    seq.block_table.extend(self.kv.allocator.allocate(1))
                except RuntimeError:
                    victim = self.running.pop()  # preempt newest
                    self._release(victim)
                    victim.status = SeqStatus.WAITING
                    victim.preemptions += 1
                    self.waiting.appendleft(victim)
                    if victim is seq:
                        continue
                    i = min(i, len(self.running))
                    continue
            i += 1
        if not self.running:
            return None
        n = len(self.running)
        bs = self.kv.block_size
        toks = np.empty(n, dtype=np.int64)
        poss = np.empty(n, dtype=np.int64)
        slots = np.empty(n, dtype=np.int64)
        lens: list[int] = []
        for i, seq in enumerate(self.running):
            pos = seq.num_tokens - 1
            out = seq.output_token_ids
            toks[i] = out[-1] if out else seq.prompt_token_ids[-1]
            poss[i] = pos
            slots[i] = seq.block_table[pos // bs] * bs + pos % bs
            lens.append(pos + 1)
        batch.seqs = list(self.running)
        batch.token_ids = toks
        batch.positions = poss
        batch.slot_mapping = slots
        batch.seq_lens = lens
        return batch

    # -- lifecycle ---------------------------------------------------------
    def on_prefill_done(self, batch: ScheduledBatch) -> None:
        for seq in batch.seqs:
            seq.num_cached_tokens = seq.num_tokens
            self.running.append(seq)

    def finish_seq(self, seq: Sequence, reason: str) -> None:
        self._release(seq)
        seq.finish(reason)
        if seq in self.running:
            self.running.remove(seq)

    def _release(self, seq: Sequence) -> None:
        if seq.block_table:
            self.kv.allocator.free(seq.block_table)
            seq.block_table = []
