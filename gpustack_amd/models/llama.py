"""Llama-family decoder (Llama 2/3, Qwen2.x, Qwen3 dense) on the native ops.

Weights live as fused bf16 parameters shaped for serving GEMMs
(qkv / gate_up fused, hipBLASLt via F.linear), attention + norms + rope +
activation run on the hand-written CDNA4 kernels (gpustack_amd.ops), and
tensor parallelism shards heads/intermediate across the node's xGMI-linked
GPUs with RCCL all-reduce after o_proj and down_proj.

Replaces the engine-side model code the reference delegates to vLLM
(SURVEY.md §2.9 #1). Covers ModelSpec.attention_bias (Qwen2) and
ModelSpec.qk_norm (Qwen3).
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops
from ..engine.config import EngineConfig, ModelSpec
from ..parallel import Communicator


@dataclass
class ForwardMeta:
    """Per-step tensors describing the batch (built by the model runner)."""

    is_prefill: bool
    positions: torch.Tensor       # [T] int64
    slot_mapping: torch.Tensor    # [T] int64
    logits_indices: torch.Tensor  # [N] int64 rows of x to project to logits
    # prefill
    seq_lens_list: list[int] | None = None
    tile_start: torch.Tensor | None = None
    tile_q0: torch.Tensor | None = None
    tile_len: torch.Tensor | None = None
    # decode
    block_tables: torch.Tensor | None = None  # [N, maxb] int32
    seq_lens: torch.Tensor | None = None      # [N] int32
    # mixed batches: rows [0, num_prefill_tokens) are prefill, the rest are
    # single-token decode rows
    num_prefill_tokens: int = 0
    # prefix-cache suffix / chunked-prefill continuation rows running through
    # the paged prefill-with-history kernel: (tiles 5-tuple, starts, hists,
    # news); block_tables is then per-SEQ ([nseq, maxb]) instead of per-row
    suffix_meta: tuple | None = None
    # dynamic multi-LoRA: per-batch adapter row groups (models/lora.py
    # BatchLora); None when no row in the batch uses an adapter
    lora: object | None = None
    # prefill context parallelism (parallel/cp.py CPMeta): this rank runs a
    # subset of the batch rows; K/V are all-gathered across the CP group
    # before the cache write so every rank's paged cache holds the FULL
    # sequences (slot_mapping is then the full-batch mapping, positions
    # stay local-row)
    cp: object | None = None


class W4Pack:
    """Runtime-packed int4 weight (models/quantized.py pack_w4_runtime):
    qw u8 [N, K/2] frag-ordered, sc/zs bf16 [N, K/128]."""

    __slots__ = ("qw", "sc", "zs", "shape")

    def __init__(self, qw, sc, zs):
        self.qw, self.sc, self.zs = qw, sc, zs
        self.shape = (qw.shape[0], qw.shape[1] * 2)


def qlinear(x, w, pack: "W4Pack | None", bias=None):
    """F.linear with an optional W4 runtime pack.

    Default GPU path: the w4_dequant HIP kernel expands the packed weight
    into a transient bf16 tensor (allocator-cached) and hipBLASLt runs the
    GEMM at full MFMA rate — weights stay RESIDENT packed (the capacity
    goal) at a measured ~2.25x packed-read traffic per use. The in-register
    dequant MFMA kernel (w4_gemm) is kept behind GPUSTACK_AMD_W4_KERNEL=1:
    both its LDS-staged and direct-L2 variants measured slower than
    dequant+hipBLASLt at serving shapes so far (profiles/r04)."""
    if pack is None:
        return F.linear(x, w, bias)
    if x.is_cuda:
        import os as _os

        hip = ops._load_hip()
        if (x.shape[0] <= 1024
                and _os.environ.get("GPUSTACK_AMD_W4_KERNEL", "0") == "1"):
            out = torch.empty(x.shape[0], pack.shape[0], dtype=x.dtype,
                              device=x.device)
            hip.w4_gemm(out, x.contiguous(), pack.qw, pack.sc, pack.zs)
            if bias is not None:
                out += bias
            return out
        wt = torch.empty(pack.shape, dtype=torch.bfloat16, device=x.device)
        hip.w4_dequant(wt, pack.qw, pack.sc, pack.zs)
        return F.linear(x, wt, bias)
    from .quantized import dequant_w4_runtime

    wt = dequant_w4_runtime(pack.qw, pack.sc, pack.zs).to(x.dtype)
    return F.linear(x, wt, bias)


class Attention(nn.Module):
    def __init__(self, spec: ModelSpec, tp_size: int, comm: Communicator,
                 dtype, layer_idx: int = 0):
        super().__init__()
        self.spec = spec
        self.comm = comm
        self.hq = spec.num_heads // tp_size
        self.hkv = max(1, spec.num_kv_heads // tp_size)
        self.d = spec.head_dim
        # GLM-style partial rotary: only the first rot_dim dims rotate
        self.rot_dim = int(spec.head_dim * spec.partial_rotary_factor)
        self.scale = spec.attn_scale or self.d ** -0.5
        self.softcap = spec.attn_logit_softcap
        self.hq_full = spec.num_heads
        # SmolLM3 NoPE layers skip rotary entirely
        self.use_rope = (spec.no_rope_layers is None
                         or layer_idx >= len(spec.no_rope_layers)
                         or bool(spec.no_rope_layers[layer_idx]))
        self.hkv_full = max(spec.num_kv_heads, tp_size) \
            if spec.num_kv_heads < tp_size else spec.num_kv_heads
        h = spec.hidden_size
        qkv_out = (self.hq + 2 * self.hkv) * self.d
        self.qkv_w = nn.Parameter(torch.empty(qkv_out, h, dtype=dtype), requires_grad=False)
        self.qkv_b = (
            nn.Parameter(torch.empty(qkv_out, dtype=dtype), requires_grad=False)
            if spec.attention_bias else None
        )
        self.o_w = nn.Parameter(torch.empty(h, self.hq * self.d, dtype=dtype), requires_grad=False)
        self.o_b = (nn.Parameter(torch.empty(h, dtype=dtype),
                                 requires_grad=False)
                    if spec.o_proj_bias else None)
        # GPT-OSS: per-head sink logits (TP-sharded with the q heads) and
        # per-layer sliding window (layer_types or even-layer default)
        self.sinks = (nn.Parameter(torch.empty(self.hq, dtype=torch.float32),
                                   requires_grad=False)
                      if spec.attention_sinks else None)
        if spec.sliding_window > 0:
            if spec.layer_types is not None and layer_idx < len(spec.layer_types):
                sliding = spec.layer_types[layer_idx] == "sliding_attention"
            else:
                sliding = layer_idx % 2 == 0  # GPT-OSS default alternation
            self.window = spec.sliding_window if sliding else 0
        else:
            self.window = 0
        self.qkv_pack: W4Pack | None = None   # W4 runtime (qlinear)
        self.o_pack: W4Pack | None = None
        self.layer_idx = 0  # set by LlamaForCausalLM
        nq, nk = self.hq * self.d, self.hkv * self.d
        self._qkv_projs = [("q_proj", 0, nq), ("k_proj", nq, nk),
                           ("v_proj", nq + nk, nk)]
        self._o_projs = [("o_proj", 0, h)]
        if spec.qk_norm:
            if spec.qk_norm_full:
                # OLMo-2: norm over the FULL projection (sharded weight
                # slice per rank; the mean reduces across the tp group)
                self.q_norm = nn.Parameter(
                    torch.empty(self.hq * self.d, dtype=dtype),
                    requires_grad=False)
                self.k_norm = nn.Parameter(
                    torch.empty(self.hkv * self.d, dtype=dtype),
                    requires_grad=False)
            else:
                self.q_norm = nn.Parameter(torch.empty(self.d, dtype=dtype), requires_grad=False)
                self.k_norm = nn.Parameter(torch.empty(self.d, dtype=dtype), requires_grad=False)

    def forward(self, x, meta: ForwardMeta, cos_sin, k_cache, v_cache):
        T = x.shape[0]
        qkv = qlinear(x, self.qkv_w, self.qkv_pack, self.qkv_b)
        if meta.lora is not None:
            qkv = qkv.contiguous()
            meta.lora.apply(self.layer_idx, x, qkv, self._qkv_projs)
        nq, nk = self.hq * self.d, self.hkv * self.d
        # strided views into the fused buffer — the HIP kernels take row
        # strides, so no layout copies on the hot path
        q = qkv[:, :nq].unflatten(1, (self.hq, self.d))
        k = qkv[:, nq:nq + nk].unflatten(1, (self.hkv, self.d))
        v = qkv[:, nq + nk:].unflatten(1, (self.hkv, self.d))
        if self.spec.qk_norm and not self.spec.qk_norm_after_rope:
            q = q.contiguous()
            k = k.contiguous()
            v = v.contiguous()
            if self.spec.qk_norm_full:
                self._full_qk_norm(q, k)
            else:
                ops.rms_norm(q.view(-1, self.d), q.view(-1, self.d), self.q_norm, self.spec.rms_norm_eps)
                ops.rms_norm(k.view(-1, self.d), k.view(-1, self.d), self.k_norm, self.spec.rms_norm_eps)
        if self.use_rope:
            ops.rotary_embedding(meta.positions, q, k, cos_sin, self.d,
                                 self.rot_dim, mode=self.spec.rope_mode)
        if self.spec.qk_norm and self.spec.qk_norm_after_rope:
            # Hunyuan-dense: per-head norm on the ROTATED q/k
            q = q.contiguous()
            k = k.contiguous()
            v = v.contiguous()
            ops.rms_norm(q.view(-1, self.d), q.view(-1, self.d),
                         self.q_norm, self.spec.rms_norm_eps)
            ops.rms_norm(k.view(-1, self.d), k.view(-1, self.d),
                         self.k_norm, self.spec.rms_norm_eps)
        if meta.cp is not None:
            # CP prefill: assemble the full-batch K/V (global position
            # order) so the cache write below covers every chunk, not just
            # this rank's rows; q/attention stay local-row
            kw, vw = meta.cp.gather_kv(k.contiguous(), v.contiguous())
        else:
            kw, vw = k, v
        ops.reshape_and_cache(kw, vw, k_cache, v_cache, meta.slot_mapping)
        out = torch.empty(T, self.hq, self.d, dtype=q.dtype, device=q.device)
        sw = {}
        if self.sinks is not None or self.window:
            sw = {"sinks": self.sinks, "window": self.window}
        if self.softcap:
            sw["softcap"] = self.softcap
        if meta.is_prefill:
            tp = meta.num_prefill_tokens or T
            ops.varlen_prefill_attn(
                out[:tp], q[:tp], k[:tp], v[:tp], meta.seq_lens_list, self.scale,
                tiles=(meta.tile_start, meta.tile_q0, meta.tile_len), **sw,
            )
            if tp < T:  # mixed: decode rows ride the same forward
                ops.paged_attn_decode(
                    out[tp:], q[tp:], k_cache, v_cache,
                    meta.block_tables, meta.seq_lens, self.scale, **sw,
                )
        elif meta.suffix_meta is not None:
            tiles, starts, hists, news = meta.suffix_meta
            ops.paged_prefill_attn(out, q, k_cache, v_cache,
                                   meta.block_tables, starts, hists, news,
                                   self.scale, tiles=tiles, **sw)
        else:
            ops.paged_attn_decode(
                out, q, k_cache, v_cache, meta.block_tables, meta.seq_lens,
                self.scale, **sw
            )
        o = qlinear(out.view(T, -1), self.o_w, self.o_pack, self.o_b)
        if meta.lora is not None:
            meta.lora.apply(self.layer_idx, out.view(T, -1), o, self._o_projs)
        return self.comm.all_reduce(o)

    def _full_qk_norm(self, q, k):
        """OLMo-2 full-projection RMSNorm: the mean square runs over ALL
        num_heads*head_dim dims; under TP each rank holds a shard, so the
        per-row sum-of-squares all-reduces over the tp group before the
        (sharded) weight scales — bit-compatible with the single-rank
        math up to summation order."""
        import torch.distributed as dist

        eps = self.spec.rms_norm_eps
        for t, w, full in ((q, self.q_norm, self.hq_full * self.d),
                           (k, self.k_norm, self.hkv_full * self.d)):
            T = t.shape[0]
            flat = t.view(T, -1).float()
            ss = flat.pow(2).sum(-1, keepdim=True)
            if self.comm.tp_size > 1:
                dist.all_reduce(ss, group=self.comm.group)
            inv = torch.rsqrt(ss / full + eps)
            t.view(T, -1).copy_((flat * inv * w.float()).to(t.dtype))


class MLAAttention(nn.Module):
    """Multi-head Latent Attention (DeepSeek V2/V3/R1 — reference serves
    these via vLLM; SURVEY.md §2.10 model families). MI355X-first design:
    the PAGED CACHE stores only the compressed latent per token
    ([kv_lora_rank + qk_rope_head_dim] = 576 f.ex. — ~57x smaller than
    expanded MHA KV), and attention runs in the ABSORBED formulation:

        score(t, l) = (q_nope W_uk) · c_l  +  q_rope · k_rope_l
        out(t)      = (softmax · C) W_uv

    which is algebraically exact vs HF's expand-then-attend and never
    materializes per-head K/V. The latent projections (kv_a) are
    REPLICATED across TP ranks (each rank writes the same 576-wide cache
    — deterministic by construction) while q_b / kv_b / o shard by head.
    CPU path is the serving oracle (HF-logits-exact,
    tests/test_deepseek.py); the CDNA4 absorbed-decode kernel is the r3
    item and this module fails loudly on CUDA until it lands.
    """

    def __init__(self, spec: ModelSpec, tp_size: int, comm: Communicator,
                 dtype, layer_idx: int = 0):
        super().__init__()
        self.spec = spec
        self.comm = comm
        self.layer_idx = layer_idx
        h = spec.hidden_size
        assert spec.num_heads % tp_size == 0
        self.nh = spec.num_heads // tp_size
        self.dn = spec.qk_nope_head_dim
        self.dr = spec.qk_rope_head_dim
        self.dq = self.dn + self.dr
        self.dv = spec.v_head_dim
        self.r = spec.kv_lora_rank
        mk = lambda *shape: nn.Parameter(torch.empty(*shape, dtype=dtype),
                                         requires_grad=False)
        if spec.q_lora_rank:
            self.q_a_w = mk(spec.q_lora_rank, h)
            self.q_a_norm = mk(spec.q_lora_rank)
            self.q_b_w = mk(self.nh * self.dq, spec.q_lora_rank)
            self.q_w = None
        else:
            self.q_w = mk(self.nh * self.dq, h)
            self.q_a_w = self.q_a_norm = self.q_b_w = None
        self.kv_a_w = mk(self.r + self.dr, h)       # replicated
        self.kv_a_norm = mk(self.r)
        self.kv_b_w = mk(self.nh * (self.dn + self.dv), self.r)
        self.o_w = mk(h, self.nh * self.dv)
        # softmax scale: qk_head_dim^-0.5, yarn-mscale-adjusted exactly as
        # HF yarn_apply_mscale (modeling_deepseek_v3.py)
        scale = self.dq ** -0.5
        rs = spec.rope_scaling or {}
        if rs.get("rope_type", rs.get("type", "default")) != "default":
            msd = rs.get("mscale_all_dim", 0)
            if msd:
                import math as _math

                f = rs["factor"]
                m = 1.0 if f <= 1 else 0.1 * msd * _math.log(f) + 1.0
                scale = scale * m * m
        self.scale = scale

    def _rope(self, x, cs):
        """x [T, ..., dr], cs [T, dr] = (cos_half, sin_half). DeepSeek
        checkpoints store the rotary dims INTERLEAVED (pairs (x0,x1));
        HF's interleave rope emits the rotated evens then odds — output
        layout is consistent between q and k so scores match."""
        half = self.dr // 2
        shape = [x.shape[0]] + [1] * (x.dim() - 2) + [half]
        cos = cs[:, :half].view(shape)
        sin = cs[:, half:].view(shape)
        if self.spec.rope_interleave:
            x1, x2 = x[..., 0::2], x[..., 1::2]
            return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], -1)
        x1, x2 = x[..., :half], x[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], -1)

    def forward(self, x, meta: ForwardMeta, cos_sin, k_cache, v_cache):
        import os as _os

        T = x.shape[0]
        # mixed prefill batches (decode rows riding along) keep the loud
        # default: the per-seq tables in meta cover prefill seqs only
        pure = (not meta.is_prefill
                or (meta.block_tables is not None
                    and (meta.num_prefill_tokens or T) == T))
        gpu_ok = (x.is_cuda
                  and _os.environ.get("GPUSTACK_AMD_MLA_KERNEL") == "1"
                  and self.r == 512 and self.dr == 64 and self.nh % 16 == 0
                  and pure)
        if x.is_cuda and not gpu_ok:
            raise NotImplementedError(
                "MLA (DeepSeek) GPU serving is gated until the r3 kernel "
                "validation pass (set GPUSTACK_AMD_MLA_KERNEL=1; requires "
                "kv_lora_rank 512 / qk_rope 64 / local heads %16==0) — "
                "the CPU path is the HF-exact oracle it verifies against")
        spec = self.spec
        if self.q_w is not None:
            q = F.linear(x, self.q_w)
        else:
            qa = F.linear(x, self.q_a_w)
            qn = torch.empty_like(qa)
            ops.rms_norm(qn, qa, self.q_a_norm, spec.rms_norm_eps)
            q = F.linear(qn, self.q_b_w)
        q = q.view(T, self.nh, self.dq)
        cs = cos_sin[meta.positions]
        q_nope, q_rot = q[..., :self.dn], q[..., self.dn:]
        q_rot = self._rope(q_rot, cs)
        lat = F.linear(x, self.kv_a_w)              # [T, r + dr]
        c = torch.empty(T, self.r, dtype=x.dtype, device=x.device)
        ops.rms_norm(c, lat[:, :self.r].contiguous(), self.kv_a_norm,
                     spec.rms_norm_eps)
        k_rot = self._rope(lat[:, self.r:], cs)
        lat_rows = torch.cat([c, k_rot], dim=-1)    # [T, r + dr]
        # cache write: one latent row per token ([nblocks, 1, BS, r+dr])
        slots = meta.slot_mapping
        valid = slots >= 0
        flat = k_cache.view(-1, self.r + self.dr)
        if bool(valid.all()):
            flat[slots] = lat_rows.to(flat.dtype)
        elif bool(valid.any()):
            flat[slots[valid]] = lat_rows[valid].to(flat.dtype)
        # absorbed projections
        kvb = self.kv_b_w.view(self.nh, self.dn + self.dv, self.r)
        uk = kvb[:, :self.dn]                       # [nh, dn, r]
        uv = kvb[:, self.dn:]                       # [nh, dv, r]
        if x.is_cuda and meta.is_prefill:
            # gated EXPAND-prefill: prefill is compute-bound, so expand
            # per-head K/V from the latent (two hipBLASLt einsums) and run
            # the MFMA flash kernel at D_qk=192 over D_v=128 values — full
            # matrix-core rate, no O(T*L) re-reads
            k_full = torch.cat([
                torch.einsum("tr,hdr->thd", c.float(), uk.float()),
                k_rot.float().unsqueeze(1).expand(T, self.nh, self.dr),
            ], dim=-1).to(x.dtype).contiguous()
            v_full = torch.einsum("tr,hdr->thd", c.float(),
                                  uv.float()).to(x.dtype).contiguous()
            q192 = torch.cat([q_nope.float(), q_rot.float()],
                             dim=-1).to(x.dtype).contiguous()
            out_pf = torch.empty(T, self.nh, self.dv, dtype=x.dtype,
                                 device=x.device)
            ops.varlen_prefill_attn(out_pf, q192, k_full, v_full,
                                    meta.seq_lens_list, self.scale)
            o = F.linear(out_pf.reshape(T, self.nh * self.dv), self.o_w)
            return self.comm.all_reduce(o)
        # absorbed projection of q (decode/suffix/CPU paths only — the
        # expand-prefill branch above never needs it)
        q_lat = torch.einsum("thd,hdr->thr", q_nope.float(), uk.float())
        if x.is_cuda:  # gated absorbed kernel path (decode/suffix rows)
            # suffix rows run ROW-WISE (len = abs position + 1 through the
            # seq's block table) — correct, with O(T*L) latent re-reads;
            # the tiled absorbed-suffix kernel is the r3 follow-up.
            bt, lens = self._gpu_row_tables(meta, T, x.device)
            q_cat = torch.cat([q_lat.to(x.dtype), q_rot.to(x.dtype)],
                              dim=-1).contiguous()
            ctx_lat = torch.empty(T, self.nh, self.r, dtype=torch.float32,
                                  device=x.device)
            ops.mla_decode(ctx_lat, q_cat, k_cache, bt, lens, self.scale)
            o = torch.einsum("nhr,hdr->nhd", ctx_lat, uv.float())
            o = F.linear(o.to(x.dtype).reshape(T, self.nh * self.dv),
                         self.o_w)
            return self.comm.all_reduce(o)
        out = torch.empty(T, self.nh, self.dv, dtype=x.dtype, device=x.device)
        for rows, ctx, qpos in self._segments(meta, lat_rows, k_cache):
            C = ctx.float()                         # [L, r+dr]
            ql = q_lat[rows]                        # [n, nh, r]
            qr = q_rot[rows].float()                # [n, nh, dr]
            scores = (torch.einsum("nhr,lr->nhl", ql, C[:, :self.r])
                      + torch.einsum("nhd,ld->nhl", qr, C[:, self.r:]))
            scores = scores * self.scale
            L = C.shape[0]
            kvpos = torch.arange(L, device=x.device)
            mask = kvpos.view(1, 1, L) <= qpos.view(-1, 1, 1)
            scores = scores.masked_fill(~mask, float("-inf"))
            probs = torch.softmax(scores, dim=-1)
            ctx_lat = torch.einsum("nhl,lr->nhr", probs, C[:, :self.r])
            o = torch.einsum("nhr,hdr->nhd", ctx_lat, uv.float())
            out[rows] = o.to(x.dtype)
        o = F.linear(out.reshape(T, self.nh * self.dv), self.o_w)
        return self.comm.all_reduce(o)

    def _gpu_row_tables(self, meta: ForwardMeta, T: int, dev):
        """Per-ROW (block_table, context_len) tensors for the absorbed
        kernel across the three batch shapes."""
        if meta.is_prefill:
            # per-seq tables in meta.block_tables (rows [0, n_pre));
            # mixed decode tails reuse their own decode tables — the
            # prefill _meta packs per-seq tables for ALL seqs here
            reps, lens = [], []
            for i, L in enumerate(meta.seq_lens_list or []):
                reps.extend([i] * L)
                lens.extend(range(1, L + 1))
            bt = meta.block_tables[torch.tensor(reps, device=dev)]
            return bt, torch.tensor(lens, dtype=torch.int32, device=dev)
        if meta.suffix_meta is not None:
            _tiles, starts, hists, news = meta.suffix_meta
            reps, lens = [], []
            for i, (h, n) in enumerate(zip(hists, news)):
                reps.extend([i] * n)
                lens.extend(range(h + 1, h + n + 1))
            bt = meta.block_tables[torch.tensor(reps, device=dev)]
            return bt, torch.tensor(lens, dtype=torch.int32, device=dev)
        return meta.block_tables, meta.seq_lens

    def _segments(self, meta: ForwardMeta, lat_rows, k_cache):
        """Yield (row_indices, latent_context [L, r+dr], qpos [n]) per
        sequence for the three batch shapes. Positions are absolute; the
        latent context is position-ordered from 0."""
        dev = lat_rows.device
        D = self.r + self.dr
        flat = k_cache.view(-1, D)
        bs = self.spec_block_size(k_cache)
        if meta.is_prefill:
            off = 0
            for L in (meta.seq_lens_list or []):
                rows = torch.arange(off, off + L, device=dev)
                yield rows, lat_rows[off:off + L], torch.arange(L, device=dev)
                off += L
            tp = meta.num_prefill_tokens or lat_rows.shape[0]
            if tp < lat_rows.shape[0]:  # mixed: decode rows ride along
                for j in range(lat_rows.shape[0] - tp):
                    L = int(meta.seq_lens[j])
                    bt = meta.block_tables[j]
                    idx = (bt[torch.arange(L, device=dev) // bs].long() * bs
                           + torch.arange(L, device=dev) % bs)
                    yield (torch.tensor([tp + j], device=dev), flat[idx],
                           torch.tensor([L - 1], device=dev))
        elif meta.suffix_meta is not None:
            _tiles, starts, hists, news = meta.suffix_meta
            for i, (st, hist, n) in enumerate(zip(starts, hists, news)):
                L = hist + n
                bt = meta.block_tables[i]
                idx = (bt[torch.arange(L, device=dev) // bs].long() * bs
                       + torch.arange(L, device=dev) % bs)
                yield (torch.arange(st, st + n, device=dev), flat[idx],
                       torch.arange(hist, L, device=dev))
        else:
            for j in range(lat_rows.shape[0]):
                L = int(meta.seq_lens[j])
                bt = meta.block_tables[j]
                idx = (bt[torch.arange(L, device=dev) // bs].long() * bs
                       + torch.arange(L, device=dev) % bs)
                yield (torch.tensor([j], device=dev), flat[idx],
                       torch.tensor([L - 1], device=dev))

    @staticmethod
    def spec_block_size(k_cache):
        return k_cache.shape[2]


class MLP(nn.Module):

    def __init__(self, spec: ModelSpec, tp_size: int, comm: Communicator, dtype):
        super().__init__()
        self.comm = comm
        h = spec.hidden_size
        self.i = spec.intermediate_size // tp_size
        self.no_gate = spec.mlp_no_gate  # Arcee: down(act(up(x)))
        gu_rows = self.i if self.no_gate else 2 * self.i
        self.gate_up_w = nn.Parameter(torch.empty(gu_rows, h, dtype=dtype), requires_grad=False)
        self.down_w = nn.Parameter(torch.empty(h, self.i, dtype=dtype), requires_grad=False)
        self.gate_up_pack: W4Pack | None = None  # W4 runtime (qlinear)
        self.down_pack: W4Pack | None = None
        self.layer_idx = 0  # set by LlamaForCausalLM
        self._gu_projs = ([("up_proj", 0, self.i)] if self.no_gate else
                          [("gate_proj", 0, self.i), ("up_proj", self.i, self.i)])
        self._down_projs = [("down_proj", 0, h)]
        self.act = spec.mlp_act

    def forward(self, x, meta: ForwardMeta | None = None):
        gu = qlinear(x, self.gate_up_w, self.gate_up_pack)
        if meta is not None and meta.lora is not None:
            meta.lora.apply(self.layer_idx, x, gu, self._gu_projs)
        if self.no_gate:
            # Arcee: single up projection, relu^2 activation
            act = (torch.square(F.relu(gu)) if self.act == "relu2"
                   else F.silu(gu))
        elif self.act == "gelu_tanh":
            # Gemma GeGLU (HF gelu_pytorch_tanh); torch pointwise path —
            # the fused CDNA4 geglu variant is r3 (same slot as the
            # GPT-OSS clamped-swiglu kernel work)
            act = (F.gelu(gu[:, :self.i], approximate="tanh")
                   * gu[:, self.i:])
        else:
            act = torch.empty(x.shape[0], self.i, dtype=x.dtype,
                              device=x.device)
            ops.silu_and_mul(act, gu)
        down = (qlinear(act, self.down_w, self.down_pack)
                if self.down_pack is not None
                else ops.linear_auto(act, self.down_w))
        if meta is not None and meta.lora is not None:
            meta.lora.apply(self.layer_idx, act, down, self._down_projs)
        return self.comm.all_reduce(down)


class MoEMLP(nn.Module):
    """Router + expert bank (Qwen3-MoE `Qwen3MoeSparseMoeBlock`, Mixtral).

    Experts live as stacked fused tensors [E, 2*I_loc, h] / [E, h, I_loc];
    TP shards each expert's intermediate dim (same all-reduce pattern as
    the dense MLP). Dispatch is sort-free eager: per-expert index_select +
    index_add over the top-k assignment table — a fused grouped-GEMM
    kernel is the round-2 optimization; this path is correct under graphs
    capture (fixed shapes per expert loop iteration are not required
    because MoE models currently run eager decode).
    """

    def __init__(self, spec: ModelSpec, tp_size: int, comm: Communicator, dtype):
        super().__init__()
        self.comm = comm
        self.spec = spec
        h = spec.hidden_size
        self.e = spec.num_experts
        self.top_k = spec.num_experts_per_tok
        self.i = spec.moe_intermediate_size // tp_size
        self.router_w = nn.Parameter(torch.empty(self.e, h, dtype=dtype),
                                     requires_grad=False)
        self.gate_up_w = nn.Parameter(
            torch.empty(self.e, 2 * self.i, h, dtype=dtype), requires_grad=False)
        self.down_w = nn.Parameter(
            torch.empty(self.e, h, self.i, dtype=dtype), requires_grad=False)
        # W4 runtime (weights.convert_to_w4_runtime): per-expert packs;
        # bank-shaped consumers dequant a TRANSIENT bf16 bank per call
        # (resident capacity is the point: Qwen3-235B W4 fits one GPU)
        self.gate_up_packs: list | None = None
        self.down_packs: list | None = None
        # GLM/DeepSeek-style extensions (spec.router_mode "sigmoid_bias"):
        # learned correction bias on the routing scores and a SHARED dense
        # expert applied to every token (TP shards its intermediate)
        if spec.router_mode == "sigmoid_bias" or spec.router_logit_bias:
            self.router_bias = nn.Parameter(
                torch.zeros(self.e, dtype=torch.float32), requires_grad=False)
        else:
            self.router_bias = None
        # GPT-OSS expert biases (TP: gate_up bias shards with the
        # intermediate; the down bias is added on tp rank 0 only so the
        # all-reduce sums it exactly once)
        if spec.moe_bias:
            self.gate_up_b = nn.Parameter(
                torch.zeros(self.e, 2 * self.i, dtype=dtype),
                requires_grad=False)
            self.down_b = nn.Parameter(
                torch.zeros(self.e, h, dtype=dtype), requires_grad=False)
        else:
            self.gate_up_b = None
            self.down_b = None
        self.shared_gate_w = (
            nn.Parameter(torch.empty(1, h, dtype=dtype),
                         requires_grad=False)
            if spec.shared_expert_gated else None)
        if spec.n_shared_experts > 0:
            si = spec.moe_intermediate_size * spec.n_shared_experts // tp_size
            self.shared_i = si
            self.shared_gate_up_w = nn.Parameter(
                torch.empty(2 * si, h, dtype=dtype), requires_grad=False)
            self.shared_down_w = nn.Parameter(
                torch.empty(h, si, dtype=dtype), requires_grad=False)
        else:
            self.shared_gate_up_w = None

    def _routing_consts(self, T: int, device):
        """Per-T cached routing constants — rebuilt tensors per layer per
        step showed up as ~27% glue kernels in the r2 profile.

        Entries are kept for the module's lifetime in a dict: a hipGraph
        captured after an eager warmup references the cached tensor by
        ADDRESS, so evicting/overwriting an entry would leave captured
        graphs reading freed (re-used) memory — garbage token indices and
        out-of-bounds gathers at replay. Only decode-bucket-sized T values
        are cached (<=1024 seqs); large prefill shapes compute fresh."""
        cache = getattr(self, "_rc", None)
        if cache is None:
            cache = self._rc = {}
        c = cache.get(T)
        if c is None:
            flat_tok = torch.arange(T, device=device).repeat_interleave(self.top_k)
            ones = torch.ones(T * self.top_k, dtype=torch.int32, device=device)
            if T > 1024:
                return flat_tok, ones
            c = cache[T] = (flat_tok, ones)
        return c

    def forward(self, x, meta: ForwardMeta | None = None):
        # MoE expert MLPs do not take LoRA (rejected at adapter load);
        # attention adapters still apply upstream
        T = x.shape[0]
        rw = getattr(self, "_router_w_f32", None)
        if rw is None or rw.device != x.device:
            self._router_w_f32 = self.router_w.float()
            rw = self._router_w_f32
        logits = F.linear(x.float(), rw)                          # [T, E]
        if self.spec.router_logit_bias and self.router_bias is not None:
            logits = logits + self.router_bias  # GPT-OSS: bias on logits
        if self.spec.router_mode == "sigmoid_bias":
            # GLM-4.5/DeepSeek routing (HF Glm4MoeTopkRouter): sigmoid
            # scores; the learned bias only influences the CHOICE, the
            # routing weight is the raw score; optional grouped top-k
            scores = torch.sigmoid(logits)
            choice = scores + self.router_bias
            ng, tg = self.spec.n_group, self.spec.topk_group
            if ng > 1:
                gs = (choice.view(T, ng, self.e // ng)
                      .topk(2, dim=-1)[0].sum(dim=-1))
                gi = torch.topk(gs, k=tg, dim=-1, sorted=False)[1]
                gmask = torch.zeros_like(gs).scatter_(1, gi, 1)
                smask = gmask.unsqueeze(-1).expand(T, ng, self.e // ng)                     .reshape(T, self.e)
                choice = choice.masked_fill(~smask.bool(), float("-inf"))
            experts = torch.topk(choice, k=self.top_k, dim=-1,
                                 sorted=False)[1]
            weights = scores.gather(1, experts)
            if self.spec.norm_topk_prob:
                weights = weights / (weights.sum(dim=-1, keepdim=True) + 1e-20)
            weights = weights * self.spec.routed_scaling_factor
        elif self.spec.norm_topk_prob:
            # softmax over all experts -> top-k -> renormalize == softmax
            # restricted to the top-k logits (Qwen3-MoE default, Mixtral)
            weights, experts = torch.topk(logits, self.top_k, dim=-1)
            weights = torch.softmax(weights, dim=-1)
        else:
            # un-renormalized: keep the full-softmax probabilities
            probs = torch.softmax(logits, dim=-1)
            weights, experts = torch.topk(probs, self.top_k, dim=-1)
        flat_exp = experts.reshape(-1)                            # [T*k]
        flat_tok, ones_i32 = self._routing_consts(T, x.device)
        flat_w32 = weights.reshape(-1)                            # f32
        # Per-assignment contributions land at UNIQUE rows of a [T*k, h]
        # buffer, then reduce over the fixed k axis — index_add_ over
        # duplicated token rows would use atomics, whose order (and thus
        # the bf16 rounding) varies run to run and flips the next layer's
        # router on near-ties (observed on HW).
        # Three dispatch paths:
        #  - fused grouped-GEMM HIP kernels (ops/csrc/moe_gemm.hip): no host
        #    sync anywhere, fixed launch grids — hipGraph-capturable; the
        #    default on GPU for MFMA-aligned dims
        #  - decode-shaped torch fallback: ONE padded strided-batched GEMM
        #    pair (pays a per-layer counts.max() host sync)
        #  - prefill-shaped: per-expert loop GEMMs (large per-expert work)
        # decode-shaped batches take the fused kernels (which re-stream
        # each expert's weight panel per 16-row m-tile — free when experts
        # hold <=32 rows, ruinous at prefill occupancy, where the
        # per-expert hipBLASLt loop wins). Under hipGraph CAPTURE the
        # fused path is forced for ANY size: the torch fallbacks sync with
        # the host per layer, which capture cannot record (large decode
        # buckets on many-expert models would otherwise crash at init).
        capturing = x.is_cuda and torch.cuda.is_current_stream_capturing()
        if self._fused_ok(x) and (flat_exp.numel() <= 32 * self.e
                                  or capturing):
            out = self._fused_dispatch(x, flat_exp, flat_tok, flat_w32,
                                       ones_i32)
            return self.comm.all_reduce(self._add_shared(x, out))
        flat_w = flat_w32.to(x.dtype)
        contrib = x.new_zeros(T * self.top_k, x.shape[1])
        # prefill-shaped on GPU with many experts: padded-bmm. The r2
        # profile showed the per-expert loop costing ~25% of MoE GPU time
        # in glue kernels (nonzero/index/select per expert per layer); the
        # padding overcompute (~cap/avg) is far cheaper at hipBLASLt batch
        # rates. The loop remains for few-expert models (big per-expert
        # GEMMs, negligible glue), CPU, and pathological imbalance.
        import os as _os

        # default OFF: hipBLASLt strided-batched GEMM faults at large odd
        # per-expert row counts inside the engine (reproduced at
        # [E=128, M~800, K=2048] bf16 even with 64-aligned M; standalone
        # repros at smaller caps pass) — the per-expert loop is the safe
        # prefill path until the library issue is understood
        use_bmm = (self.e >= 16 and x.is_cuda
                   and _os.environ.get("GPUSTACK_AMD_MOE_BMM", "0") == "1")
        if use_bmm:
            counts = torch.zeros(self.e, dtype=torch.long, device=x.device)
            counts.scatter_add_(0, flat_exp, torch.ones_like(flat_exp))
            cap = int(counts.max())  # one host sync per layer
            if self.e * cap > 8 * flat_exp.numel():
                use_bmm = False  # extreme imbalance: padding would explode
        if use_bmm:
            self._bmm_dispatch(x, contrib, flat_exp, flat_tok, flat_w, cap)
        else:
            self._loop_dispatch(x, contrib, flat_exp, flat_tok, flat_w)
        out = contrib.view(T, self.top_k, -1).sum(dim=1).to(x.dtype)
        return self.comm.all_reduce(self._add_shared(x, out))

    # ONE scratch per (device, shape) shared by every layer: layers
    # dequant into it sequentially on the compute stream, so a captured
    # decode graph records kernels against a STATIC address (no per-
    # bucket transient replication) and eager mode allocates it once.
    _bank_scratch: dict = {}

    def _bank(self, which: str):
        """bf16 expert bank for the fused/bmm paths: the resident tensor,
        or the shared scratch filled by dequanting this layer's per-
        expert W4 packs."""
        packs = (self.gate_up_packs if which == "gate_up"
                 else self.down_packs)
        w = self.gate_up_w if which == "gate_up" else self.down_w
        if packs is None:
            return w
        p0 = packs[0]
        dev = p0.qw.device
        key = (str(dev), self.e) + p0.shape
        bank = MoEMLP._bank_scratch.get(key)
        if bank is None:
            bank = torch.empty((self.e,) + p0.shape, dtype=torch.bfloat16,
                               device=dev)
            MoEMLP._bank_scratch[key] = bank
        if dev.type == "cuda":
            hip = ops._load_hip()
            for e, pk in enumerate(packs):
                hip.w4_dequant(bank[e], pk.qw, pk.sc, pk.zs)
        else:
            from .quantized import dequant_w4_runtime

            for e, pk in enumerate(packs):
                bank[e].copy_(dequant_w4_runtime(pk.qw, pk.sc, pk.zs))
        return bank

    def _expert_w(self, which: str, e: int):
        """Per-expert weight for the loop path (qlinear handles packs)."""
        packs = (self.gate_up_packs if which == "gate_up"
                 else self.down_packs)
        w = self.gate_up_w if which == "gate_up" else self.down_w
        if packs is None:
            return w[e], None
        return None, packs[e]

    def _add_shared(self, x, out):
        """Shared dense expert (GLM-4.5: every token, added to the routed
        mix BEFORE the TP all-reduce so one collective covers both)."""
        if self.shared_gate_up_w is None:
            return out
        gu = F.linear(x, self.shared_gate_up_w)
        act = torch.empty(x.shape[0], self.shared_i, dtype=x.dtype,
                          device=x.device)
        ops.silu_and_mul(act, gu)
        sh = F.linear(act, self.shared_down_w)
        if self.shared_gate_w is not None:
            # Qwen2-MoE: token-wise sigmoid gate on the shared expert
            # (gate is tiny [1, h] — replicated, NOT TP-sharded)
            sh = sh * torch.sigmoid(F.linear(x, self.shared_gate_w))
        return out + sh

    def _fused_ok(self, x) -> bool:
        import os

        # silu/no-bias is the GPU-validated fused configuration; the
        # GPT-OSS variant (clamped swiglu + expert biases, moe_gemm.hip
        # act_mode 1) is written + compile-checked but opt-in until the r3
        # numerics pass (same gate as the attention sinks/window kernels)
        act_ok = (self.spec.moe_act == "silu" and self.gate_up_b is None) or (
            self.spec.moe_act == "clamped_swiglu"
            and os.environ.get("GPUSTACK_AMD_OSS_KERNELS") == "1")
        return (x.is_cuda and x.dtype == torch.bfloat16
                and os.environ.get("GPUSTACK_AMD_FUSED_MOE", "1") == "1"
                and act_ok
                and self.i % 128 == 0 and x.shape[1] % 128 == 0
                and ops.hip_available())

    def _fused_dispatch(self, x, flat_exp, flat_tok, flat_w32, ones_i32):
        """Sync-free grouped expert GEMMs: sort assignments by expert on
        device, run the two fused kernels, reduce over the k axis. Every
        tensor shape here depends only on (T, k, E) — safe under hipGraph
        capture (reference capability: vLLM fused_moe; re-designed for
        CDNA4 per SURVEY.md §2.9 #1)."""
        T = x.shape[0]
        TK = flat_exp.numel()
        order = torch.argsort(flat_exp, stable=True)
        s_tok = flat_tok[order].to(torch.int32)
        # scatter_add instead of bincount: bincount computes max() on host
        counts = torch.zeros(self.e, dtype=torch.int32, device=x.device)
        counts.scatter_add_(0, flat_exp, ones_i32)
        offs = (counts.cumsum(0, dtype=torch.int32) - counts).to(torch.int32)
        hip = ops._load_hip()
        act = x.new_empty(TK, self.i)
        act_mode = 1 if self.spec.moe_act == "clamped_swiglu" else 0
        hip.moe_gate_up_silu(act, x, self._bank("gate_up"), s_tok, offs,
                             counts, bias=self.gate_up_b, act_mode=act_mode)
        contrib = x.new_empty(TK, x.shape[1])
        # down bias on tp rank 0 only: the TP all-reduce must sum it once
        db = self.down_b if self.comm.tp_rank == 0 else None
        hip.moe_down_scale(contrib, act, self._bank("down"), offs, counts,
                           order.to(torch.int32), flat_w32, bias=db)
        return contrib.view(T, self.top_k, -1).sum(dim=1).to(x.dtype)

    def _act_mul(self, gu):
        """[T', 2i] -> [T', i]: SiLU-mul (fused kernel/ref) or GPT-OSS
        clamped swiglu ((up+1) * gate*sigmoid(1.702*gate), clamps +-7).
        gu is in OUR fused layout [gate; up] (GPT-OSS checkpoints
        de-interleave at load)."""
        if self.spec.moe_act == "clamped_swiglu":
            g = gu[..., :self.i].float().clamp(max=7.0)
            u = gu[..., self.i:].float().clamp(-7.0, 7.0)
            return ((u + 1.0) * (g * torch.sigmoid(g * 1.702))).to(gu.dtype)
        act = torch.empty(gu.shape[0], self.i, dtype=gu.dtype,
                          device=gu.device)
        ops.silu_and_mul(act, gu)
        return act

    def _loop_dispatch(self, x, contrib, flat_exp, flat_tok, flat_w):
        hit = torch.bincount(flat_exp, minlength=self.e)
        tp0 = self.comm.tp_rank == 0
        for e in torch.nonzero(hit, as_tuple=False).flatten().tolist():
            rows = torch.nonzero(flat_exp == e, as_tuple=False).flatten()
            idx = flat_tok[rows]
            xe = x.index_select(0, idx)
            gw, gp = self._expert_w("gate_up", e)
            gu = qlinear(xe, gw, gp,
                         self.gate_up_b[e] if self.gate_up_b is not None
                         else None)
            act = self._act_mul(gu)
            dw, dp = self._expert_w("down", e)
            he = qlinear(act, dw, dp,
                         self.down_b[e] if (self.down_b is not None and tp0)
                         else None)
            contrib[rows] = he * flat_w[rows].unsqueeze(1)

    def _bmm_dispatch(self, x, contrib, flat_exp, flat_tok, flat_w,
                      cap: int | None = None):
        import os as _os

        dbg = _os.environ.get("GPUSTACK_AMD_MOE_DEBUG", "0") == "1"

        def _ck(tag):
            if dbg:
                torch.cuda.synchronize()
                print(f"bmm[{tag}] ok", flush=True)

        TK = flat_exp.numel()
        order = torch.argsort(flat_exp, stable=True)
        s_exp = flat_exp[order]
        s_tok = flat_tok[order]
        counts = torch.bincount(s_exp, minlength=self.e)
        if cap is None:
            cap = int(counts.max())      # one host sync per layer
        if cap == 0:
            return
        cap = (cap + 63) // 64 * 64  # GEMM-friendly M; pad rows are zero
        _ck(f"route cap={cap} TK={TK}")
        offs = counts.cumsum(0) - counts
        pos = torch.arange(TK, device=x.device) - offs[s_exp]
        xpad = x.new_zeros(self.e, cap, x.shape[1])
        xpad[s_exp, pos] = x[s_tok]
        _ck("xpad")
        gu = torch.bmm(xpad, self._bank("gate_up").transpose(1, 2))  # [E, cap, 2i]
        if self.gate_up_b is not None:
            gu = gu + self.gate_up_b.unsqueeze(1)
        _ck("bmm1")
        act = self._act_mul(gu.reshape(self.e * cap, 2 * self.i))
        _ck("silu")
        hd = torch.bmm(act.view(self.e, cap, self.i),
                       self._bank("down").transpose(1, 2))     # [E, cap, h]
        if self.down_b is not None and self.comm.tp_rank == 0:
            hd = hd + self.down_b.unsqueeze(1)
        _ck("bmm2")
        contrib[order] = hd[s_exp, pos] * flat_w[order].unsqueeze(1)
        _ck("combine")



def _layer_norm(out, x, w, eps):
    """Cohere mean-centered LayerNorm (weight only, no bias), computed in
    fp32 like HF CohereLayerNorm (modeling_cohere.py) and cast back."""
    xf = x.float()
    mu = xf.mean(-1, keepdim=True)
    var = (xf - mu).pow(2).mean(-1, keepdim=True)
    out.copy_(((xf - mu) * torch.rsqrt(var + eps) * w.float()).to(x.dtype))


class DecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, tp_size: int, comm: Communicator,
                 dtype, layer_idx: int = 0):
        super().__init__()
        self.spec = spec
        if spec.kv_lora_rank:
            self.attn = MLAAttention(spec, tp_size, comm, dtype,
                                     layer_idx=layer_idx)
        else:
            self.attn = Attention(spec, tp_size, comm, dtype,
                                  layer_idx=layer_idx)
        if spec.num_experts > 0 and layer_idx >= spec.first_k_dense_replace:
            self.mlp = MoEMLP(spec, tp_size, comm, dtype)
        else:
            # dense layer — all layers of a dense model, or the first
            # first_k_dense_replace layers of a GLM/DeepSeek-style MoE
            self.mlp = MLP(spec, tp_size, comm, dtype)
        h = spec.hidden_size
        self.input_norm = nn.Parameter(torch.empty(h, dtype=dtype), requires_grad=False)
        self.post_attn_norm = nn.Parameter(torch.empty(h, dtype=dtype), requires_grad=False)
        if spec.sandwich_norms:
            # Gemma-2: post-attn/post-ffn norms run BEFORE the residual
            # adds, plus a pre-ffn norm — four norms per layer
            self.pre_ff_norm = nn.Parameter(torch.empty(h, dtype=dtype),
                                            requires_grad=False)
            self.post_ff_norm = nn.Parameter(torch.empty(h, dtype=dtype),
                                             requires_grad=False)
        elif spec.norm_after:
            # OLMo-2: only output norms (input_norm doubles as the
            # post-attn norm slot, post_attn_norm as the post-ffn one —
            # the loader maps HF names accordingly)
            self.pre_ff_norm = self.post_ff_norm = None
        else:
            self.pre_ff_norm = self.post_ff_norm = None

    def forward(self, x, residual, meta, cos_sin, k_cache, v_cache):
        eps = self.spec.rms_norm_eps
        if self.spec.parallel_block:
            return self._forward_parallel(x, meta, cos_sin, k_cache,
                                          v_cache, eps)
        if self.spec.sandwich_norms:
            return self._forward_sandwich(x, meta, cos_sin, k_cache,
                                          v_cache, eps)
        if self.spec.norm_after:
            return self._forward_norm_after(x, meta, cos_sin, k_cache,
                                            v_cache, eps)
        if residual is None:
            residual = x
            h = torch.empty_like(x)
            ops.rms_norm(h, x, self.input_norm, eps)
        else:
            h = x
            ops.fused_add_rms_norm(h, residual, self.input_norm, eps)
        a = self.attn(h, meta, cos_sin, k_cache, v_cache)
        rm = self.spec.residual_multiplier
        if rm:  # Granite: scale sublayer outputs before the residual add
            a = a * rm
        ops.fused_add_rms_norm(a, residual, self.post_attn_norm, eps)
        m = self.mlp(a, meta)
        if rm:
            m = m * rm
        return m, residual

    def _forward_parallel(self, x, meta, cos_sin, k_cache, v_cache, eps):
        """Cohere layer flow (true hidden stream, residual sentinel
        None): one shared input LayerNorm feeds BOTH sublayers and their
        outputs add to the residual together:
            x = x + attn(ln(x)) + mlp(ln(x))"""
        h = torch.empty_like(x)
        _layer_norm(h, x, self.input_norm, eps)
        a = self.attn(h, meta, cos_sin, k_cache, v_cache)
        m = self.mlp(h, meta)
        return x + a + m, None

    def _forward_norm_after(self, x, meta, cos_sin, k_cache, v_cache, eps):
        """OLMo-2 layer flow (true hidden stream, residual sentinel None):
            x = x + norm_attn(attn(x)); x = x + norm_ff(mlp(x))
        (sublayers see the UNNORMED stream; norms scale their outputs)."""
        a = self.attn(x, meta, cos_sin, k_cache, v_cache)
        ops.rms_norm(a, a, self.input_norm, eps)
        x = x + a
        m = self.mlp(x, meta)
        ops.rms_norm(m, m, self.post_attn_norm, eps)
        return x + m, None

    def _forward_sandwich(self, x, meta, cos_sin, k_cache, v_cache, eps):
        """Gemma-2 layer flow (residual carried explicitly — the layer
        returns the true hidden stream, residual sentinel None):
            x = x + norm_post_attn(attn(norm_in(x)))
            x = x + norm_post_ff(mlp(norm_pre_ff(x)))"""
        h = torch.empty_like(x)
        ops.rms_norm(h, x, self.input_norm, eps)
        a = self.attn(h, meta, cos_sin, k_cache, v_cache)
        ops.rms_norm(a, a, self.post_attn_norm, eps)
        x = x + a
        ops.rms_norm(h, x, self.pre_ff_norm, eps)
        m = self.mlp(h, meta)
        ops.rms_norm(m, m, self.post_ff_norm, eps)
        return x + m, None


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: EngineConfig, comm: Communicator, device):
        super().__init__()
        spec = cfg.spec
        self.spec = spec
        self.cfg = cfg
        self.comm = comm
        self.device = torch.device(device)
        dtype = getattr(torch, cfg.dtype)
        self.dtype = dtype
        # pipeline partition: this rank owns layers
        # [layer_offset, layer_offset + num_local_layers); remainder layers
        # go to the EARLY stages so the last stage (which also runs the
        # lm_head GEMM and sampling) carries less per-step work
        pp, pr = comm.pp_size, comm.pp_rank
        part = getattr(cfg, "pp_partition", None)
        if part and len(part) == pp and sum(part) == spec.num_layers:
            # explicit per-stage layer counts (the native analog of the
            # reference's per-GPU GGUF tensor_split proportions,
            # gguf_resource_fit_selector.py:370-441): the scheduler sizes
            # stages proportionally to each GPU's free VRAM
            counts = list(part)
        else:
            counts = [spec.num_layers // pp
                      + (1 if i < spec.num_layers % pp else 0)
                      for i in range(pp)]
        self.layer_offset = sum(counts[:pr])
        self.num_local_layers = counts[pr]
        first, last = comm.is_first_stage, comm.is_last_stage
        if first or (last and spec.tie_word_embeddings):
            self.embed = nn.Parameter(
                torch.empty(spec.vocab_size, spec.hidden_size, dtype=dtype),
                requires_grad=False)
        else:
            self.embed = None
        self.layers = nn.ModuleList(
            [DecoderLayer(spec, cfg.tp_size, comm, dtype,
                          layer_idx=self.layer_offset + i)
             for i in range(self.num_local_layers)]
        )
        for i, layer in enumerate(self.layers):
            layer.attn.layer_idx = self.layer_offset + i
            layer.mlp.layer_idx = self.layer_offset + i
        if last:
            self.final_norm = nn.Parameter(
                torch.empty(spec.hidden_size, dtype=dtype), requires_grad=False)
            if spec.tie_word_embeddings:
                self.lm_head = self.embed
            else:
                self.lm_head = nn.Parameter(
                    torch.empty(spec.vocab_size, spec.hidden_size, dtype=dtype),
                    requires_grad=False)
        else:
            self.final_norm = None
            self.lm_head = None
        self.lm_head_pack: W4Pack | None = None  # W4 runtime (qlinear)
        self.offload = None  # CpuOffload streamer (engine/offload.py)
        if spec.kv_lora_rank:
            # MLA: rope acts on the qk_rope dims only (the latent's rope
            # slice and each head's q tail)
            cache = ops.build_cos_sin_cache(
                spec.qk_rope_head_dim, spec.qk_rope_head_dim,
                cfg.max_model_len,
                base=spec.rope_theta, scaling=spec.rope_scaling,
            )
        else:
            cache = ops.build_cos_sin_cache(
                spec.head_dim,
                int(spec.head_dim * spec.partial_rotary_factor),
                cfg.max_model_len,
                base=spec.rope_theta, scaling=spec.rope_scaling,
            )
        self.register_buffer("cos_sin", cache.to(device), persistent=False)
        if spec.rope_local_theta:
            # Gemma-3: sliding layers rope at the LOCAL base, unscaled
            local = ops.build_cos_sin_cache(
                spec.head_dim, spec.head_dim, cfg.max_model_len,
                base=spec.rope_local_theta,
            )
            self.register_buffer("cos_sin_local", local.to(device),
                                 persistent=False)
        else:
            self.cos_sin_local = None
        self.to(device)

    @torch.inference_mode()
    def forward(self, token_ids: torch.Tensor, meta: ForwardMeta, kv,
                return_hidden: bool = False, return_both: bool = False):
        if self.comm.pp_size > 1 and not self.comm.is_first_stage:
            # previous stage sends the summed residual stream; starting the
            # local stack with residual=None reproduces the single-rank
            # fused_add_rms_norm numerics exactly (it stores x+residual in
            # bf16 before normalizing)
            x = self.comm.recv_hidden(
                (token_ids.shape[0], self.spec.hidden_size),
                self.dtype, self.device)
        else:
            x = F.embedding(token_ids, self.embed)
            if self.spec.embed_scale:
                # Gemma: sqrt(hidden) embedding scale, cast like HF
                # (normalizer materialized in the embed dtype)
                x = x * torch.tensor(self.spec.embed_scale,
                                     dtype=x.dtype, device=x.device)
        residual = None
        off = self.offload
        if off is not None:
            off.begin()
        for i, layer in enumerate(self.layers):
            if off is not None and i >= off.first:
                off.bind(i)
            cs = (self.cos_sin_local
                  if (self.cos_sin_local is not None and layer.attn.window)
                  else self.cos_sin)
            x, residual = layer(x, residual, meta, cs, kv.k_caches[i], kv.v_caches[i])
        if self.comm.pp_size > 1 and not self.comm.is_last_stage:
            if residual is None:  # sandwich layers carry the true stream
                self.comm.send_hidden(x.contiguous())
            else:
                s = (x.float() + residual.float()).to(self.dtype)
                self.comm.send_hidden(s)
            return None
        if self.spec.norm_type == "layernorm":
            assert residual is None  # parallel_block carries true stream
            _layer_norm(x, x, self.final_norm, self.spec.rms_norm_eps)
        elif residual is None:
            ops.rms_norm(x, x, self.final_norm, self.spec.rms_norm_eps)
        else:
            ops.fused_add_rms_norm(x, residual, self.final_norm,
                                   self.spec.rms_norm_eps)
        if return_hidden:
            return x  # all rows, post final norm (embedding serving)
        hidden = x[meta.logits_indices]
        logits = qlinear(hidden, self.lm_head, self.lm_head_pack)
        if self.spec.final_logit_softcap:
            cap = self.spec.final_logit_softcap
            logits = torch.tanh(logits / cap) * cap
        if self.spec.logits_scaling:
            logits = logits / self.spec.logits_scaling
        if self.spec.logits_multiplier:
            logits = logits * self.spec.logits_multiplier
        if return_both:  # draft-model speculative needs the features too
            return logits, hidden
        return logits
