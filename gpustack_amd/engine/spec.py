"""Speculative decoding: n-gram (prompt-lookup) draft + verify.

Reference parity: the control plane's speculative_config (EAGLE3 / MTP /
NGRAM, gpustack/schemas/models.py:80-83,397-414 routed to engine flags at
worker/backends/vllm.py:532-566); here the engine implements the n-gram
method natively, with the draft-model methods (EAGLE-style) as a later
round.

Verify strategy (MI355X-native, graph-friendly): each decode step feeds a
FIXED window of 1+K tokens per sequence — the last sampled token plus K
draft tokens — as 1+K rows whose seq_lens increase by one. The paged
decode kernel handles multi-row queries as-is (row j attends positions
0..p+j, its own K/V freshly written), so no separate verify kernel is
needed; greedy acceptance keeps the longest matching prefix. Rejected
positions hold stale KV that is overwritten when the real token reaches
that position, so correctness is unconditional.
"""
from __future__ import annotations

from .sequence import Sequence


class NgramProposer:
    def __init__(self, num_draft_tokens: int = 3, ngram_max: int = 3, ngram_min: int = 1):
        self.k = num_draft_tokens
        self.ngram_max = ngram_max
        self.ngram_min = ngram_min

    def propose(self, seq: Sequence) -> list[int]:
        """Prompt-lookup: find the most recent earlier occurrence of the
        trailing n-gram and copy its continuation. Always returns exactly
        k tokens (padded with the last token) so batch shape is static."""
        ctx = seq.all_token_ids
        n = len(ctx)
        draft: list[int] | None = None
        for g in range(self.ngram_max, self.ngram_min - 1, -1):
            if n < g + 1:
                continue
            tail = ctx[n - g:]
            # search right-to-left, excluding the trailing match itself
            for s in range(n - g - 1, -1, -1):
                if ctx[s:s + g] == tail:
                    cont = ctx[s + g: s + g + self.k]
                    if cont:
                        draft = list(cont)
                    break
            if draft:
                break
        if draft is None:
            draft = []
        while len(draft) < self.k:
            draft.append(ctx[-1])
        return draft[: self.k]


def accept_tokens(fed_drafts: list[int], sampled: list[int]) -> list[int]:
    """Greedy acceptance: sampled[j] is the model's next token after
    position j of the fed window. Keep sampled[0]; keep sampled[j] while
    sampled[j-1] == fed_drafts[j-1]."""
    out = [sampled[0]]
    for j in range(1, len(sampled)):
        if sampled[j - 1] != fed_drafts[j - 1]:
            break
        out.append(sampled[j])
    return out
