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
    seq_lens: list[int] = field(default_factory=list)   # context length per ROW
    # speculative decoding: each seq contributes 1+K rows (fed window =
    # last token + K drafts); see engine/spec.py
    rows_per_seq: int = 1
    drafts: list[list[int]] | None = None
    k_eff: list[int] | None = None
    # mixed batches: prefill tokens first, then decode rows (one per running
    # seq) — decode piggybacks on the prefill step's weight stream
    n_prefill_seqs: int = 0
    num_prefill_tokens: int = 0
    decode_seq_lens: list[int] | None = None
    # prefix-cache suffix batch: decode-style rows, variable count per seq,
    # one sampled token per seq (engine/kv_cache.py CachingBlockAllocator)
    is_suffix: bool = False
    suffix_rows: list[int] | None = None   # rows per seq
    # chunked prefill: per prefill-seq flag — False means this step computes
    # an intermediate chunk whose sampled token must be discarded and whose
    # seq must not yet join the running set
    chunk_final: list[bool] | None = None

    @property
    def num_tokens(self) -> int:
        return len(self.token_ids)


class Scheduler:
    def __init__(self, cfg: EngineConfig, kv: KVCache):
        self.cfg = cfg
        self.kv = kv
        self.waiting: deque[Sequence] = deque()
        self.running: list[Sequence] = []
        self.swapped: deque[Sequence] = deque()  # offloaded to host DRAM
        # chunked prefill: the one sequence currently being admitted in
        # budget-sized chunks; chunk steps alternate 1:1 with decode steps
        self._chunking: Sequence | None = None
        self._chunk_decode_turn = False
        self.proposer = None
        self.spec_k = 0
        self.spec_method = None
        spec = getattr(cfg, "speculative", None)
        method = spec.get("method", "ngram") if spec else None
        if method == "ngram":
            from .spec import NgramProposer

            self.spec_method = method
            self.spec_k = int(spec.get("num_draft_tokens", 3))
            self.proposer = NgramProposer(
                self.spec_k,
                int(spec.get("ngram_max", 3)),
                int(spec.get("ngram_min", 1)),
            )
        elif method in ("eagle", "eagle3", "mtp"):
            # draft-model speculative: proposals are computed on-GPU by the
            # runner's EagleProposer each step (engine/eagle.py) and arrive
            # via seq.next_draft
            self.spec_method = method
            self.spec_k = int(spec.get("num_draft_tokens", 3))

    # -- queue ops ---------------------------------------------------------
    def add(self, seq: Sequence) -> None:
        self._insert_waiting(seq, front=False)

    def _insert_waiting(self, seq: Sequence, front: bool) -> None:
        """Priority-ordered admission (vLLM priority scheduling analog):
        higher priority enters ahead of lower; within a priority class,
        new arrivals go last and preempted resumers (front=True) first."""
        pr = seq.params.priority
        if pr == 0 and not front and (not self.waiting
                                      or self.waiting[-1].params.priority >= 0):
            self.waiting.append(seq)  # common case: plain FIFO
            return
        for i, s in enumerate(self.waiting):
            p = s.params.priority
            if (p < pr) or (front and p <= pr):
                self.waiting.insert(i, seq)
                return
        self.waiting.append(seq)

    def _pop_victim(self) -> tuple[int, Sequence]:
        """Preemption victim: lowest priority loses; ties break to the
        newest (latest-admitted) sequence."""
        idx = min(range(len(self.running)),
                  key=lambda j: (self.running[j].params.priority, -j))
        return idx, self.running.pop(idx)

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
        for s in list(self.swapped):
            if s.request_id == request_id:
                if s.host_block_table:
                    self.kv.host_allocator.free(s.host_block_table)
                    s.host_block_table = []
                s.finish("abort")
                self.swapped.remove(s)
                return True
        if (self._chunking is not None
                and self._chunking.request_id == request_id):
            s = self._chunking
            self._chunking = None
            self._release(s)
            s.finish("abort")
            return True
        return False

    @property
    def num_unfinished(self) -> int:
        return (len(self.waiting) + len(self.running) + len(self.swapped)
                + (1 if self._chunking is not None else 0))

    def has_work(self) -> bool:
        return self.num_unfinished > 0

    # -- scheduling --------------------------------------------------------
    def schedule(self) -> ScheduledBatch | None:
        self._swap_in_ready()
        if self._chunking is not None:
            # alternate chunk and decode steps so running seqs keep a
            # bounded time-between-tokens during long-prompt admission
            if self._chunk_decode_turn and self.running:
                self._chunk_decode_turn = False
                b = self._schedule_decode()
                if b is not None:
                    return b
            b = self._schedule_chunk()
            if b is not None:
                self._chunk_decode_turn = True
                return b
            return self._schedule_decode()  # KV pressure: let decode drain
        if getattr(self.cfg, "enable_prefix_caching", False):
            sb = self._schedule_suffix()
            if sb is not None:
                return sb
        batch = self._schedule_prefill()
        if batch is not None:
            import os

            # Mixed steps measured ~5% slower than keeping decode on the
            # hipGraph path at the default workload (A/B b_hyst vs
            # b_hyst_nomix) — opt-in only.
            if os.environ.get("GPUSTACK_AMD_MIXED", "0") == "1":
                self._append_decode_rows(batch)
            else:
                batch.n_prefill_seqs = len(batch.seqs)
                batch.num_prefill_tokens = batch.num_tokens
            return batch
        return self._schedule_decode()

    def _append_decode_rows(self, batch: ScheduledBatch) -> None:
        """Mixed batching: running sequences decode one token inside the
        prefill step (shared GEMM weight stream; no decode stall during
        admission bursts). Spec decoding is not applied to mixed steps."""
        batch.n_prefill_seqs = len(batch.seqs)
        batch.num_prefill_tokens = batch.num_tokens
        if not self.running:
            return
        # ensure slots exist (same preemption rules as decode-only)
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            pos = seq.num_tokens - 1
            if (pos + 1) > len(seq.block_table) * self.kv.block_size:
                try:
                    seq.block_table.extend(self.kv.allocator.allocate(1))
                except RuntimeError:
                    vidx, victim = self._pop_victim()
                    self._release(victim)
                    victim.status = SeqStatus.WAITING
                    victim.preemptions += 1
                    self._insert_waiting(victim, front=True)
                    if victim is seq:
                        continue
                    if vidx < i:
                        i -= 1
                    i = min(i, len(self.running))
                    continue
            i += 1
        bs = self.kv.block_size
        lens: list[int] = []
        for seq in self.running:
            pos = seq.num_tokens - 1
            out = seq.output_token_ids
            batch.seqs.append(seq)
            batch.token_ids.append(out[-1] if out else seq.prompt_token_ids[-1])
            batch.positions.append(pos)
            batch.slot_mapping.append(seq.block_table[pos // bs] * bs + pos % bs)
            lens.append(pos + 1)
        batch.decode_seq_lens = lens

    def _swap_in_ready(self) -> None:
        while (self.swapped
               and len(self.running) < self.cfg.max_num_seqs
               and len(self.swapped[0].host_block_table) <= self.kv.allocator.num_free):
            seq = self.swapped.popleft()
            seq.block_table = self.kv.swap_in(seq.host_block_table)
            seq.host_block_table = []
            seq.status = SeqStatus.RUNNING
            self.running.append(seq)

    def would_admit(self) -> bool:
        """True if the next schedule() may open a prefill step (shared
        predicate with _schedule_prefill; also used by the async decode
        pipeline to decide when it must drain)."""
        if not self.waiting and not self.swapped:
            return False
        if not self.running:
            return True
        import time as _time

        if not self.waiting:
            return True  # swapped re-admission
        oldest = self.waiting[0].arrival_time
        return (len(self.waiting) >= self.cfg.admission_min_seqs
                or _time.monotonic() - oldest >= self.cfg.admission_max_wait_s
                or len(self.running) < self.cfg.admission_min_seqs)

    def _cache_plan(self, seq: Sequence):
        """(cached_blocks, hashes) for the longest cached prefix chain.
        Never covers the block holding position n-1 (it will be written)."""
        from .kv_cache import block_hashes

        tokens = seq.all_token_ids
        n = len(tokens)
        bs = self.kv.block_size
        hashes = block_hashes(tokens, bs)
        max_cached = (n - 1) // bs
        blocks: list[int] = []
        for h in hashes[:max_cached]:
            b = self.kv.allocator.acquire_cached(h)
            if b is None:
                break
            blocks.append(b)
        return blocks, hashes

    def _schedule_suffix(self) -> ScheduledBatch | None:
        """Admit cache-hit prompts: acquire shared prefix blocks and run
        only the suffix tokens as decode-style rows."""
        if not self.waiting:
            return None
        if self.running and not self.would_admit():
            return None
        batch = ScheduledBatch(is_prefill=False, is_suffix=True)
        batch.suffix_rows = []
        budget = self.cfg.max_prefill_tokens
        toks: list[int] = []
        poss: list[int] = []
        slots: list[int] = []
        lens: list[int] = []
        scanned = 0
        while (self.waiting and scanned < len(self.waiting) + 8
               and len(self.running) + len(batch.seqs) < self.cfg.max_num_seqs):
            seq = self.waiting[0]
            n = seq.num_tokens
            cached_blocks, hashes = self._cache_plan(seq)
            suffix = n - len(cached_blocks) * self.kv.block_size
            if (not cached_blocks
                    or suffix > self.cfg.prefix_cache_suffix_cap):
                self.kv.allocator.free(cached_blocks)  # drop refs
                break  # head-of-line is a plain prefill; let it run first
            if batch.seqs and sum(batch.suffix_rows) + suffix > budget:
                self.kv.allocator.free(cached_blocks)
                break
            need = self.kv.blocks_needed(n) - len(cached_blocks)
            try:
                own = self.kv.allocator.allocate(need)
            except RuntimeError:
                self.kv.allocator.free(cached_blocks)
                break
            self.waiting.popleft()
            scanned += 1
            seq.block_table = cached_blocks + own
            seq.block_hashes = hashes
            seq.num_cached_tokens = len(cached_blocks) * self.kv.block_size
            seq.status = SeqStatus.RUNNING
            start = seq.num_cached_tokens
            tokens = seq.all_token_ids
            batch.seqs.append(seq)
            batch.suffix_rows.append(suffix)
            toks.extend(tokens[start:])
            poss.extend(range(start, n))
            slots.extend(self.kv.slots_for(seq.block_table, start, suffix))
            lens.extend(range(start + 1, n + 1))  # row r attends 0..start+r
        if not batch.seqs:
            return None
        batch.token_ids = toks
        batch.positions = poss
        batch.slot_mapping = slots
        batch.seq_lens = lens
        batch.n_prefill_seqs = len(batch.seqs)
        return batch

    def _schedule_prefill(self) -> ScheduledBatch | None:
        if not self.waiting:
            return None
        if self.running and not self.would_admit():
            return None  # let decode keep its graph cadence
        batch = ScheduledBatch(is_prefill=True)
        budget = self.cfg.max_prefill_tokens
        while self.waiting and len(self.running) + len(batch.seqs) < self.cfg.max_num_seqs:
            seq = self.waiting[0]
            n = seq.num_tokens  # prompt + any generated tokens (preempted seqs)
            if (not batch.seqs and n > budget
                    and getattr(self.cfg, "enable_chunked_prefill", False)):
                # chunk 0: prefill only the first budget tokens; the rest
                # continues through _schedule_chunk (paged-decode rows)
                nblocks = self.kv.blocks_needed(budget)
                if nblocks > self.kv.allocator.num_free:
                    break
                self.waiting.popleft()
                seq.block_table = self.kv.allocator.allocate(nblocks)
                seq.num_cached_tokens = budget
                if getattr(self.cfg, "enable_prefix_caching", False):
                    from .kv_cache import block_hashes

                    # registered by on_prefill_done at the FINAL chunk
                    seq.block_hashes = block_hashes(seq.all_token_ids,
                                                    self.kv.block_size)
                else:
                    seq.block_hashes = None
                seq.status = SeqStatus.RUNNING
                self._chunking = seq
                self._chunk_decode_turn = True
                tokens = seq.all_token_ids
                batch.seqs.append(seq)
                batch.token_ids.extend(tokens[:budget])
                batch.positions.extend(range(budget))
                batch.slot_mapping.extend(
                    self.kv.slots_for(seq.block_table, 0, budget))
                batch.seq_lens.append(budget)
                batch.chunk_final = [False]
                return batch  # chunk steps run exclusively
            if batch.seqs and batch.num_tokens + n > budget:
                break
            nblocks = self.kv.blocks_needed(n)
            if nblocks > self.kv.allocator.num_free:
                break
            self.waiting.popleft()
            seq.block_table = self.kv.allocator.allocate(nblocks)
            seq.num_cached_tokens = 0
            seq.status = SeqStatus.RUNNING
            if getattr(self.cfg, "enable_prefix_caching", False):
                from .kv_cache import block_hashes

                seq.block_hashes = block_hashes(seq.all_token_ids,
                                                self.kv.block_size)
            tokens = seq.all_token_ids
            batch.seqs.append(seq)
            batch.token_ids.extend(tokens)
            batch.positions.extend(range(n))
            batch.slot_mapping.extend(self.kv.slots_for(seq.block_table, 0, n))
            batch.seq_lens.append(n)
        if not batch.seqs:
            return None
        return batch

    def _schedule_chunk(self) -> ScheduledBatch | None:
        """Continue the in-progress chunked prefill: the next budget-sized
        window of prompt tokens runs as paged-decode rows (the same path
        prefix-cache suffixes use — each row attends to all prior KV)."""
        seq = self._chunking
        p = seq.num_cached_tokens
        n = seq.num_tokens
        take = min(self.cfg.max_prefill_tokens, n - p)
        need = self.kv.blocks_needed(p + take) - len(seq.block_table)
        if need > 0:
            try:
                seq.block_table.extend(self.kv.allocator.allocate(need))
            except RuntimeError:
                return None  # pool exhausted; decode steps may free blocks
        batch = ScheduledBatch(is_prefill=False, is_suffix=True)
        tokens = seq.all_token_ids
        batch.seqs = [seq]
        batch.suffix_rows = [take]
        batch.token_ids = tokens[p:p + take]
        batch.positions = list(range(p, p + take))
        batch.slot_mapping = self.kv.slots_for(seq.block_table, p, take)
        batch.seq_lens = list(range(p + 1, p + take + 1))
        batch.n_prefill_seqs = 1
        final = p + take == n
        batch.chunk_final = [final]
        seq.num_cached_tokens = p + take
        if final:
            self._chunking = None
        return batch

    def _schedule_decode(self) -> ScheduledBatch | None:
        if not self.running:
            return None
        batch = ScheduledBatch(is_prefill=False)
        # Ensure every running seq has a slot for its next token; preempt from
        # the back (most recent) on pool exhaustion.
        k = self.spec_k
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            pos = seq.num_tokens - 1          # position of the token to feed
            k_eff = min(k, self.cfg.max_model_len - 1 - pos)
            k_eff = max(0, k_eff)
            if (pos + 1 + k_eff) > len(seq.block_table) * self.kv.block_size:
                need = (pos + 1 + k_eff + self.kv.block_size - 1) // self.kv.block_size \
                    - len(seq.block_table)
                try:
                    seq.block_table.extend(self.kv.allocator.allocate(need))
                except RuntimeError:
                    vidx, victim = self._pop_victim()
                    if self.kv.can_swap_out(len(victim.block_table)):
                        # offload tier: swap KV to pinned host DRAM instead
                        # of recompute
                        victim.host_block_table = self.kv.swap_out(victim.block_table)
                        victim.block_table = []
                        victim.status = SeqStatus.WAITING
                        victim.swap_outs += 1
                        self.swapped.append(victim)
                    else:
                        self._release(victim)
                        victim.status = SeqStatus.WAITING
                        victim.preemptions += 1
                        self._insert_waiting(victim, front=True)
                    if victim is seq:
                        continue
                    if vidx < i:
                        i -= 1
                    i = min(i, len(self.running))
                    continue
            i += 1
        if not self.running:
            return None
        n = len(self.running)
        bs = self.kv.block_size
        rps = 1 + self.spec_k
        toks = np.empty(n * rps, dtype=np.int64)
        poss = np.empty(n * rps, dtype=np.int64)
        slots = np.empty(n * rps, dtype=np.int64)
        lens: list[int] = []
        drafts: list[list[int]] = []
        k_effs: list[int] = []
        for i, seq in enumerate(self.running):
            pos = seq.num_tokens - 1
            out = seq.output_token_ids
            if seq.pending_tokens:
                last = 0  # placeholder: async path feeds tokens from device
            else:
                last = out[-1] if out else seq.prompt_token_ids[-1]
            if rps == 1:
                toks[i] = last
                poss[i] = pos
                slots[i] = seq.block_table[pos // bs] * bs + pos % bs
                lens.append(pos + 1)
                continue
            k_eff = max(0, min(self.spec_k, self.cfg.max_model_len - 1 - pos))
            if seq.params.spec_safe and k_eff > 0 and self.proposer is not None:
                draft = self.proposer.propose(seq)
            elif seq.params.spec_safe and k_eff > 0 and seq.next_draft:
                draft = (list(seq.next_draft) + [last] * self.spec_k)[:self.spec_k]
                seq.next_draft = None
            else:
                draft = [last] * self.spec_k
                k_eff = 0
            fed = [last] + draft
            for j in range(rps):
                r = i * rps + j
                toks[r] = fed[j]
                valid = j <= k_eff
                pj = pos + j if valid else pos + k_eff
                poss[r] = pj
                # overflow rows skip the KV write (slot -1) so they cannot
                # clobber live positions
                slots[r] = (seq.block_table[pj // bs] * bs + pj % bs) if valid else -1
                lens.append(pj + 1)
            drafts.append(draft)
            k_effs.append(k_eff)
        batch.seqs = list(self.running)
        batch.token_ids = toks
        batch.positions = poss
        batch.slot_mapping = slots
        batch.seq_lens = lens
        batch.rows_per_seq = rps
        if rps > 1:
            batch.drafts = drafts
            batch.k_eff = k_effs
        return batch

    # -- lifecycle ---------------------------------------------------------
    def on_prefill_done(self, batch: ScheduledBatch) -> None:
        n = batch.n_prefill_seqs or len(batch.seqs)
        caching = getattr(self.cfg, "enable_prefix_caching", False)
        finals = batch.chunk_final
        for i, seq in enumerate(batch.seqs[:n]):
            if finals is not None and i < len(finals) and not finals[i]:
                continue  # intermediate chunk: seq stays with the scheduler
            seq.num_cached_tokens = seq.num_tokens
            if caching and getattr(seq, "block_hashes", None):
                # register this prompt's full blocks (minus the one holding
                # position n-1, which decode will write) for later sharing
                limit = (seq.num_tokens - 1) // self.kv.block_size
                for i, h in enumerate(seq.block_hashes[:limit]):
                    self.kv.allocator.register(seq.block_table[i], h)
                seq.block_hashes = None
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
