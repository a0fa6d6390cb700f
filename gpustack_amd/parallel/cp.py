"""Prefill context parallelism (CP) — the reference's vLLM
``--prefill-context-parallel-size`` dimension (pcp is
modeled as a world-size multiplier and delegated to the engine; here the
engine is first-party, so the mechanism is too).

MI355X-first design: prefill is the compute-bound phase (MFMA flash
kernels at high arithmetic intensity), so CP splits each prompt's ROWS
into ``cp`` contiguous position chunks — rank r computes chunk r through
every layer (QKV, attention, MLP all ~1/cp the FLOPs). Attention needs
K/V for all positions ≤ the rank's rows, so each layer all-gathers the
rope'd K/V rows across the CP group (one bf16 all-gather of
T×Hkv_local×D per layer over xGMI — GQA makes this small relative to the
hidden-state traffic TP would move), writes the FULL sequence K/V into
the rank's own paged cache, and runs the rank's rows through the existing
prefill-with-history path (``hist`` = the chunk's start position). After
prefill every rank holds the complete KV cache, so decode steps run
replicated and deterministic on all ranks — the same lockstep-scheduler
contract TP followers already obey. The tail chunk (which produces the
logits row) always lands on cp_rank cp-1 (floor-bound partition), which
broadcasts sampled ids world-wide exactly like the PP sampling stage.
"""
from __future__ import annotations

import torch


def cp_bounds(length: int, cp: int) -> list[int]:
    """Chunk boundaries [b_0..b_cp] with b_i = i*L//cp: balanced contiguous
    chunks; chunk cp-1 is never empty for L >= 1 (tail ownership is fixed)."""
    return [i * length // cp for i in range(cp + 1)]


class CPMeta:
    """Per-batch CP prefill metadata.

    perm maps each GLOBAL row (seq-major, ascending position) to its flat
    index in the concatenated padded gather buffer ``rank * pad_rows +
    local_row`` — applying it to the all-gathered K/V restores full
    position order regardless of how many rows each rank contributed.
    """

    def __init__(self, comm, pad_rows: int, perm: torch.Tensor):
        self.comm = comm
        self.pad_rows = pad_rows
        self.perm = perm

    def gather_kv(self, k: torch.Tensor, v: torch.Tensor):
        """[T_local, Hkv, D] -> [T_full, Hkv, D] in global position order."""
        pad = self.pad_rows - k.shape[0]
        if pad:
            zk = k.new_zeros((pad,) + tuple(k.shape[1:]))
            k = torch.cat([k, zk], dim=0)
            v = torch.cat([v, zk.clone()], dim=0)
        kf = self.comm.cp_all_gather_rows(k)
        vf = self.comm.cp_all_gather_rows(v)
        return kf[self.perm], vf[self.perm]


def build_cp_prefill(seq_lens: list[int], cp: int, cp_rank: int):
    """Partition a pure-prefill batch's rows across the CP group.

    Returns (local_rows, hists, news, perm, pad_rows, counts):
      local_rows — global row indices this rank computes (seq-major order)
      hists[i]   — seq i's chunk start for this rank (history length for
                   the prefill-with-history attention path)
      news[i]    — seq i's chunk length on this rank (0 allowed)
      perm       — [T_total] gather permutation (see CPMeta)
      pad_rows   — per-rank padded row count for the equal-shape all-gather
      counts     — per-rank real row counts (diagnostics / dummy handling)
    """
    counts = [0] * cp
    local_rows: list[int] = []
    hists: list[int] = []
    news: list[int] = []
    # first pass: per-rank row counts (perm needs every rank's layout)
    bounds = [cp_bounds(L, cp) for L in seq_lens]
    for b in bounds:
        for r in range(cp):
            counts[r] += b[r + 1] - b[r]
    pad_rows = max(1, max(counts))
    # second pass: local rows + perm
    perm = torch.empty(sum(seq_lens), dtype=torch.long)
    offs = [0] * cp          # next local row per rank
    row = 0
    for s, (L, b) in enumerate(zip(seq_lens, bounds)):
        for r in range(cp):
            n = b[r + 1] - b[r]
            if r == cp_rank:
                hists.append(b[r])
                news.append(n)
                local_rows.extend(range(row + b[r], row + b[r + 1]))
            for j in range(n):
                perm[row + b[r] + j] = r * pad_rows + offs[r] + j
            offs[r] += n
        row += L
    return local_rows, hists, news, perm, pad_rows, counts
