"""EAGLE-style draft-model speculative decoding (method "eagle"/"eagle3").

Reference parity: speculative_config methods EAGLE3 / MTP / NGRAM
(gpustack/schemas/models.py:80-83,397-414, routed to engine flags at
worker/backends/vllm.py:532-566). The reference delegates the draft-verify
loop to vLLM; here it is first-party on the same CDNA4 kernel set:

  * draft = fc(concat(embed(token_p), target_hidden_{p-1})) -> one
    llama DecoderLayer -> norm, sharing the target's embedding, rope table
    and lm_head (random-init in this offline build; checkpoint loading of
    published EAGLE heads is a round-2 item).
  * the draft keeps its OWN 1-layer paged KV pool; draft rows reuse the
    spec-row mechanism (paged_attn_decode over rows with increasing
    seq_lens — engine/spec.py), so no separate verify kernel exists:
    drafting AND verification both run through the standard decode path.
  * per engine step: (1) extend the draft KV with the accepted tokens
    (one batched var-row decode forward, conditioned on the target
    hiddens the verify step just produced), (2) chain K greedy
    micro-steps through the draft layer to propose the next window.
  * verification is the existing 1+K-row target step + greedy
    `accept_tokens` — output is therefore IDENTICAL to plain greedy
    decoding regardless of draft quality (tested), and draft quality
    only affects speed.

Scratch positions past the accepted length hold stale draft KV that the
next extension overwrites, mirroring the target-side spec-row contract.
"""
from __future__ import annotations

import logging
from dataclasses import dataclass, field

import torch
import torch.nn.functional as F

from .. import ops
from ..models.llama import DecoderLayer, ForwardMeta
from .config import EngineConfig
from .kv_cache import BlockAllocator
from .scheduler import ScheduledBatch
from .sequence import Sequence

logger = logging.getLogger(__name__)


@dataclass
class _DraftState:
    block_table: list[int] = field(default_factory=list)
    pos: int = -1  # last position whose draft-KV entry is confirmed


class EagleProposer:
    def __init__(self, cfg: EngineConfig, model, comm, device, num_blocks: int):
        self.cfg = cfg
        self.model = model          # target model (embed/lm_head/cos_sin shared)
        self.comm = comm
        self.device = torch.device(device)
        self.k = int(cfg.speculative.get("num_draft_tokens", 3))
        spec = cfg.spec
        self.spec = spec
        self.bs = cfg.block_size
        dtype = model.dtype
        kv_heads = max(1, spec.num_kv_heads // cfg.tp_size)
        shape = (num_blocks, kv_heads, cfg.block_size, spec.head_dim)
        self.k_cache = torch.zeros(shape, dtype=dtype, device=device)
        self.v_cache = torch.zeros(shape, dtype=dtype, device=device)
        self.allocator = BlockAllocator(num_blocks)
        self.states: dict[str, _DraftState] = {}

        h = spec.hidden_size
        # "eagle"/"eagle3": ONE draft head chained k times.
        # "mtp": k chained heads with distinct weights, head m predicting
        # draft position m (DeepSeek-style multi-token prediction modules,
        # sharing one draft KV — an engineering simplification vs per-head
        # KV streams; greedy acceptance keeps outputs exact regardless).
        method = cfg.speculative.get("method", "eagle")
        self.n_heads = self.k if method == "mtp" else 1
        self.fc_ws: list[torch.Tensor] = []
        self.layers: list[DecoderLayer] = []
        self.norms: list[torch.Tensor] = []
        for _ in range(self.n_heads):
            self.fc_ws.append(torch.empty(h, 2 * h, dtype=dtype, device=device))
            self.layers.append(DecoderLayer(spec, cfg.tp_size, comm, dtype).to(device))
            self.norms.append(torch.empty(h, dtype=dtype, device=device))
        self._init_weights()
        draft_dir = cfg.speculative.get("draft_dir")
        if draft_dir:
            n = self.load_draft(draft_dir)
            logger.info("loaded draft checkpoint %s (%d tensors)", draft_dir, n)

    def _init_weights(self) -> None:
        """Deterministic TP-consistent random init (models/weights.py
        discipline: full tensors generated, then sliced per rank)."""
        from ..models.weights import _gen

        cfg, spec = self.cfg, self.spec
        tp, rank = cfg.tp_size, cfg.tp_rank
        d = spec.head_dim
        hq, hkv = spec.num_heads // tp, max(1, spec.num_kv_heads // tp)
        i_loc = spec.intermediate_size // tp
        dt, dev, seed = self.fc_ws[0].dtype, self.device, cfg.seed
        for hi in range(self.n_heads):
            pfx = f"draft{hi}"
            self.fc_ws[hi].copy_(_gen((spec.hidden_size, 2 * spec.hidden_size),
                                      f"{pfx}.fc", seed, dt, dev))
            self.norms[hi].fill_(1.0)
            layer = self.layers[hi]
            la = layer.attn
            q = _gen((spec.num_heads * d, spec.hidden_size), f"{pfx}.q", seed, dt, dev)
            k = _gen((spec.num_kv_heads * d, spec.hidden_size), f"{pfx}.k", seed, dt, dev)
            v = _gen((spec.num_kv_heads * d, spec.hidden_size), f"{pfx}.v", seed, dt, dev)
            la.qkv_w.copy_(torch.cat([
                q[rank * hq * d:(rank + 1) * hq * d],
                k[rank * hkv * d:(rank + 1) * hkv * d],
                v[rank * hkv * d:(rank + 1) * hkv * d]]))
            if la.qkv_b is not None:
                la.qkv_b.zero_()
            o = _gen((spec.hidden_size, spec.num_heads * d), f"{pfx}.o", seed, dt, dev)
            la.o_w.copy_(o[:, rank * hq * d:(rank + 1) * hq * d])
            if spec.qk_norm:
                la.q_norm.fill_(1.0)
                la.k_norm.fill_(1.0)
            gate = _gen((spec.intermediate_size, spec.hidden_size), f"{pfx}.gate", seed, dt, dev)
            up = _gen((spec.intermediate_size, spec.hidden_size), f"{pfx}.up", seed, dt, dev)
            layer.mlp.gate_up_w.copy_(torch.cat([
                gate[rank * i_loc:(rank + 1) * i_loc],
                up[rank * i_loc:(rank + 1) * i_loc]]))
            down = _gen((spec.hidden_size, spec.intermediate_size), f"{pfx}.down", seed, dt, dev)
            layer.mlp.down_w.copy_(down[:, rank * i_loc:(rank + 1) * i_loc])
            layer.input_norm.fill_(1.0)
            layer.post_attn_norm.fill_(1.0)

    def load_draft(self, draft_dir) -> int:
        """Load a published EAGLE/MTP draft checkpoint (speculative_config
        `draft_dir`; reference: spec-decode model refs routed to engine
        flags at worker/backends/vllm.py:532-566).

        Accepts the EAGLE release naming (`fc.weight`,
        `layers.0.self_attn.q_proj.weight`, optional `norm.weight`, with or
        without a `model.` prefix); head hi of an MTP checkpoint reads
        `layers.{hi}.*`. Tensors the checkpoint lacks keep their random
        init (missing norms default to ones via init). Returns the number
        of tensors loaded.
        """
        from pathlib import Path

        from safetensors import safe_open

        draft_dir = Path(draft_dir)
        files = sorted(draft_dir.glob("*.safetensors"))
        if not files:
            raise FileNotFoundError(f"no safetensors under {draft_dir}")
        tensors: dict[str, torch.Tensor] = {}
        for f in files:
            with safe_open(str(f), framework="pt") as sf:
                for name in sf.keys():
                    tensors[name] = sf.get_tensor(name)

        cfg, spec = self.cfg, self.spec
        tp, rank = cfg.tp_size, cfg.tp_rank
        d = spec.head_dim
        hq, hkv = spec.num_heads // tp, max(1, spec.num_kv_heads // tp)
        i_loc = spec.intermediate_size // tp
        dt, dev = self.fc_ws[0].dtype, self.device

        def find(*frags: str) -> torch.Tensor | None:
            for key, t in tensors.items():
                if all(f in key for f in frags):
                    return t.to(dt)
            return None

        loaded = 0

        def copy_rows(dst, t, off, rows):
            nonlocal loaded
            dst.copy_(t[off:off + rows].to(dev))
            loaded += 1

        for hi in range(self.n_heads):
            li = f"layers.{hi}." if self.n_heads > 1 else "layers.0."
            fc = find("fc.weight") if hi == 0 or self.n_heads == 1 \
                else find(f"{hi}.", "fc.weight")
            if fc is not None:
                self.fc_ws[hi].copy_(fc.to(dev))
                loaded += 1
            nrm = None
            for cand in (f"{li}norm.weight", "norm.weight", "model.norm.weight"):
                if cand in tensors:
                    nrm = tensors[cand].to(dt)
                    break
            if nrm is not None and nrm.numel() == spec.hidden_size:
                self.norms[hi].copy_(nrm.to(dev))
                loaded += 1
            layer = self.layers[hi]
            la = layer.attn
            q = find(li, "q_proj.weight")
            k = find(li, "k_proj.weight")
            v = find(li, "v_proj.weight")
            if q is not None and k is not None and v is not None:
                la.qkv_w.copy_(torch.cat([
                    q[rank * hq * d:(rank + 1) * hq * d],
                    k[rank * hkv * d:(rank + 1) * hkv * d],
                    v[rank * hkv * d:(rank + 1) * hkv * d]]).to(dev))
                loaded += 3
            o = find(li, "o_proj.weight")
            if o is not None:
                la.o_w.copy_(o[:, rank * hq * d:(rank + 1) * hq * d].to(dev))
                loaded += 1
            g, u = find(li, "gate_proj.weight"), find(li, "up_proj.weight")
            if g is not None and u is not None:
                layer.mlp.gate_up_w.copy_(torch.cat([
                    g[rank * i_loc:(rank + 1) * i_loc],
                    u[rank * i_loc:(rank + 1) * i_loc]]).to(dev))
                loaded += 2
            dn = find(li, "down_proj.weight")
            if dn is not None:
                layer.mlp.down_w.copy_(
                    dn[:, rank * i_loc:(rank + 1) * i_loc].to(dev))
                loaded += 1
            inorm = find(li, "input_layernorm.weight")
            if inorm is not None:
                layer.input_norm.copy_(inorm.to(dev))
                loaded += 1
            pnorm = find(li, "post_attention_layernorm.weight")
            if pnorm is not None:
                layer.post_attn_norm.copy_(pnorm.to(dev))
                loaded += 1
        if loaded == 0:
            raise ValueError(f"{draft_dir}: no recognizable draft tensors")
        return loaded

    # -- draft-KV bookkeeping ---------------------------------------------

    def drop(self, seq: Sequence) -> None:
        st = self.states.pop(seq.request_id, None)
        if st is not None and st.block_table:
            self.allocator.free(st.block_table)
        seq.next_draft = None

    def _ensure_blocks(self, st: _DraftState, max_pos: int) -> bool:
        need = (max_pos + self.bs) // self.bs - len(st.block_table)
        if need > 0:
            try:
                st.block_table.extend(self.allocator.allocate(need))
            except RuntimeError:
                return False
        return True

    def _slot(self, st: _DraftState, pos: int) -> int:
        return st.block_table[pos // self.bs] * self.bs + pos % self.bs

    # -- draft forward -----------------------------------------------------

    @torch.inference_mode()
    def _forward(self, tokens, h_prev, positions, slots, seq_lens, block_tables,
                 head: int = 0):
        """One draft pass in spec-row decode mode; returns draft hidden."""
        dev = self.device
        e = F.embedding(tokens, self.model.embed)
        x = F.linear(torch.cat([e, h_prev], dim=-1), self.fc_ws[head])
        max_pos = self.cfg.max_model_len - 1
        meta = ForwardMeta(
            is_prefill=False,
            positions=positions.clamp(max=max_pos),
            slot_mapping=slots,
            logits_indices=torch.arange(x.shape[0], dtype=torch.long, device=dev),
            block_tables=block_tables,
            seq_lens=seq_lens,
        )
        x, residual = self.layers[head](x, None, meta, self.model.cos_sin,
                                        self.k_cache, self.v_cache)
        ops.fused_add_rms_norm(x, residual, self.norms[head], self.spec.rms_norm_eps)
        return x

    @torch.inference_mode()
    def _forward_prefill(self, tokens, h_prev, positions, slots, lens):
        dev = self.device
        e = F.embedding(tokens, self.model.embed)
        x = F.linear(torch.cat([e, h_prev], dim=-1), self.fc_ws[0])
        tiles = ops.build_prefill_tiles(lens, dev)
        meta = ForwardMeta(
            is_prefill=True,
            positions=positions,
            slot_mapping=slots,
            logits_indices=torch.zeros(1, dtype=torch.long, device=dev),
            seq_lens_list=lens,
            tile_start=tiles[0], tile_q0=tiles[1], tile_len=tiles[2],
        )
        x, residual = self.layers[0](x, None, meta, self.model.cos_sin,
                                     self.k_cache, self.v_cache)
        ops.fused_add_rms_norm(x, residual, self.norms[0], self.spec.rms_norm_eps)
        return x

    def _argmax_token(self, hidden) -> torch.Tensor:
        from ..models.llama import qlinear

        # routes through the W4 runtime pack when lm_head is packed
        return qlinear(hidden, self.model.lm_head,
                       self.model.lm_head_pack).argmax(dim=-1)

    def _bt_tensor(self, states: list[_DraftState], rows_per: list[int]):
        maxb = max(len(st.block_table) for st in states)
        bt = torch.zeros(sum(rows_per), maxb, dtype=torch.int32)
        r = 0
        for st, n in zip(states, rows_per):
            row = torch.tensor(st.block_table, dtype=torch.int32)
            for _ in range(n):
                bt[r, :len(st.block_table)] = row
                r += 1
        return bt.to(self.device)

    @torch.inference_mode()
    def _propose_chain(self, seqs, states, h_chain, feed_tok,
                       feed_pos: list[int]) -> None:
        """Greedy micro-steps producing exactly k draft tokens.

        feed_tok: [n] tokens to feed first (seed path: the target-sampled
        token after the prompt), or None when argmax(h_chain) already IS
        the first draft (decode path: the extension's last row consumed
        the newest accepted token). feed_pos[i] = position of the first
        chained row for seq i."""
        n = len(seqs)
        dev = self.device
        drafts = []
        if feed_tok is None:
            tok = self._argmax_token(h_chain)
            drafts.append(tok)
        else:
            tok = feed_tok
        bt = self._bt_tensor(states, [1] * n)
        m = 0
        while len(drafts) < self.k:
            positions = torch.tensor([feed_pos[i] + m for i in range(n)],
                                     dtype=torch.long, device=dev)
            slots = torch.tensor(
                [self._slot(states[i], min(feed_pos[i] + m,
                                           len(states[i].block_table) * self.bs - 1))
                 for i in range(n)], dtype=torch.long, device=dev)
            seq_lens = torch.tensor([feed_pos[i] + m + 1 for i in range(n)],
                                    dtype=torch.int32, device=dev)
            head = min(len(drafts), self.n_heads - 1)  # MTP: head per position
            h_chain = self._forward(tok, h_chain, positions, slots, seq_lens,
                                    bt, head=head)
            tok = self._argmax_token(h_chain)
            drafts.append(tok)
            m += 1
        mat = torch.stack(drafts, dim=1).cpu().tolist()   # [n, k]
        for seq, row in zip(seqs, mat):
            seq.next_draft = [int(t) for t in row]

    # -- engine hooks ------------------------------------------------------

    @torch.inference_mode()
    def seed_from_prefill(self, batch: ScheduledBatch, hidden_all,
                          sampled: list[int], alive: list[bool]) -> None:
        """After a pure-prefill target step: build the draft KV over each
        prompt (inputs: token_p + target hidden_{p-1}), feed the
        target-sampled first token, and propose the first K-token window."""
        n = batch.n_prefill_seqs or len(batch.seqs)
        seqs = batch.seqs[:n]
        lens = batch.seq_lens[:n]
        dev = self.device
        states = []
        ok_seqs, ok_lens, offs, feed = [], [], [], []
        off = 0
        for i, (seq, L) in enumerate(zip(seqs, lens)):
            self.drop(seq)
            if not alive[i]:
                off += L
                continue
            st = _DraftState()
            if not self._ensure_blocks(st, L + self.k + 1):
                if st.block_table:
                    self.allocator.free(st.block_table)
                off += L
                continue
            # the sampled token's row (position L) is fed in the chain
            # below with target-confirmed identity -> confirmed through L
            st.pos = L
            self.states[seq.request_id] = st
            states.append(st)
            ok_seqs.append(seq)
            ok_lens.append(L)
            offs.append(off)
            feed.append(sampled[i])
            off += L
        if not ok_seqs:
            return
        tokens, h_rows, positions, slots = [], [], [], []
        h = hidden_all.shape[-1]
        zero = torch.zeros(1, h, dtype=hidden_all.dtype, device=dev)
        for st, seq, L, o in zip(states, ok_seqs, ok_lens, offs):
            tokens.extend(seq.prompt_token_ids[:L])
            h_rows.append(zero)
            if L > 1:
                h_rows.append(hidden_all[o:o + L - 1])
            positions.extend(range(L))
            slots.extend(self._slot(st, p) for p in range(L))
        tokens_t = torch.tensor(tokens, dtype=torch.long, device=dev)
        h_prev = torch.cat(h_rows, dim=0)
        pos_t = torch.tensor(positions, dtype=torch.long, device=dev)
        slot_t = torch.tensor(slots, dtype=torch.long, device=dev)
        out = self._forward_prefill(tokens_t, h_prev, pos_t, slot_t, ok_lens)
        last_idx = torch.tensor(
            [sum(ok_lens[:i + 1]) - 1 for i in range(len(ok_lens))],
            dtype=torch.long, device=dev)
        feed_t = torch.tensor(feed, dtype=torch.long, device=dev)
        self._propose_chain(ok_seqs, states, out[last_idx], feed_t,
                            [L for L in ok_lens])

    @torch.inference_mode()
    def step(self, batch: ScheduledBatch, hidden_rows, emitted: list[list[int]],
             alive: list[bool]) -> None:
        """After a verified decode step: extend each seq's draft KV with its
        accepted tokens (conditioned on the verify step's target hiddens)
        and propose the next window."""
        rps = batch.rows_per_seq
        dev = self.device
        states, ok_seqs, ext_tokens, ext_h, ext_pos, ext_slots = [], [], [], [], [], []
        ext_lens, rows_per, first_pos = [], [], []
        for i, seq in enumerate(batch.seqs):
            if not alive[i]:
                continue
            st = self.states.get(seq.request_id)
            P = int(batch.positions[i * rps])
            if st is None or st.pos != P:
                if st is not None:
                    self.drop(seq)
                else:
                    seq.next_draft = None
                continue
            em = emitted[i]
            j = len(em) - 1
            if not self._ensure_blocks(st, P + 1 + j + self.k + 1):
                self.drop(seq)
                continue
            for m, tok in enumerate(em):
                ext_tokens.append(tok)
                ext_pos.append(P + 1 + m)
                ext_slots.append(self._slot(st, P + 1 + m))
                ext_lens.append(P + 2 + m)
            ext_h.append(hidden_rows[i * rps:i * rps + j + 1])
            st.pos = P + 1 + j
            states.append(st)
            ok_seqs.append(seq)
            rows_per.append(j + 1)
            first_pos.append(P + 2 + j)
        if not ok_seqs:
            return
        tokens_t = torch.tensor(ext_tokens, dtype=torch.long, device=dev)
        h_prev = torch.cat(ext_h, dim=0)
        pos_t = torch.tensor(ext_pos, dtype=torch.long, device=dev)
        slot_t = torch.tensor(ext_slots, dtype=torch.long, device=dev)
        len_t = torch.tensor(ext_lens, dtype=torch.int32, device=dev)
        bt = self._bt_tensor(states, rows_per)
        out = self._forward(tokens_t, h_prev, pos_t, slot_t, len_t, bt)
        last_idx = torch.tensor(
            [sum(rows_per[:i + 1]) - 1 for i in range(len(rows_per))],
            dtype=torch.long, device=dev)
        self._propose_chain(ok_seqs, states, out[last_idx], None, first_pos)
