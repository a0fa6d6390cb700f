"""Weight init / loading for the native engine.

Two paths (reference parity: the worker's model-file manager downloads HF
safetensors — gpustack/worker/model_file_manager.py:59; here loading is
first-party):

  * random init — deterministic per (seed, tensor) and TP-consistent: full
    tensors are generated then sliced to the local shard, so TP=N shards
    always compose to the same TP=1 weights (bench uses this: no network).
  * safetensors  — maps HF checkpoint names onto the fused serving layout.
"""
from __future__ import annotations

import json
from pathlib import Path

import torch

from ..engine.config import EngineConfig


def _gen(shape, seed_key: str, base_seed: int, dtype, device, std=0.02):
    import zlib

    seed = (base_seed * 1000003 + zlib.crc32(seed_key.encode())) % (2**63)
    dev = torch.device(device)
    if dev.type == "cuda":  # device-side randn: 8B init in seconds
        g = torch.Generator(device=dev)
        g.manual_seed(seed)
        t = torch.randn(*shape, generator=g, dtype=torch.float32, device=dev) * std
        return t.to(dtype)
    g = torch.Generator(device="cpu")
    g.manual_seed(seed)
    t = torch.randn(*shape, generator=g, dtype=torch.float32) * std
    return t.to(dtype).to(device)


def _kv_slice(full, rank: int, tp: int, n_kv: int, d: int):
    """KV shard rows for `rank`: plain slice while tp <= num_kv_heads;
    beyond that, KV heads are REPLICATED — groups of tp/num_kv_heads ranks
    share one head (standard practice for high-TP GQA, e.g. TP8 over a
    4-KV-head model like Qwen3-235B-A22B)."""
    if tp <= n_kv:
        per = n_kv // tp
        return full[rank * per * d:(rank + 1) * per * d]
    repl = tp // n_kv
    head = rank // repl
    return full[head * d:(head + 1) * d]


def random_init(model, cfg: EngineConfig) -> None:
    spec = model.spec
    tp, rank = cfg.tp_size, cfg.tp_rank
    dtype, device = model.dtype, model.device
    seed = cfg.seed
    hq, hkv, d = spec.num_heads // tp, max(1, spec.num_kv_heads // tp), spec.head_dim
    i_loc = spec.intermediate_size // tp
    off = getattr(model, "layer_offset", 0)  # pipeline stage-local layers

    if model.embed is not None:
        model.embed.copy_(_gen((spec.vocab_size, spec.hidden_size), "embed", seed, dtype, device))
    if model.lm_head is not None and not spec.tie_word_embeddings:
        model.lm_head.copy_(_gen((spec.vocab_size, spec.hidden_size), "lm_head", seed, dtype, device))
    if model.final_norm is not None:
        model.final_norm.fill_(1.0)

    for local_i, layer in enumerate(model.layers):
        li = off + local_i
        if spec.kv_lora_rank:
            _random_init_mla_attn(layer, spec, li, seed, dtype, device, tp,
                                  rank)
            if hasattr(layer.mlp, "router_w"):
                _random_init_moe_layer(layer, spec, li, seed, dtype, device,
                                       tp, rank)
            else:
                up = _gen((spec.intermediate_size, spec.hidden_size),
                          f"{li}.up", seed, dtype, device)
                if getattr(layer.mlp, "no_gate", False):
                    layer.mlp.gate_up_w.copy_(
                        up[rank * i_loc:(rank + 1) * i_loc])
                else:
                    gate = _gen((spec.intermediate_size, spec.hidden_size),
                                f"{li}.gate", seed, dtype, device)
                    layer.mlp.gate_up_w.copy_(torch.cat([
                        gate[rank * i_loc:(rank + 1) * i_loc],
                        up[rank * i_loc:(rank + 1) * i_loc],
                    ]))
                down = _gen((spec.hidden_size, spec.intermediate_size),
                            f"{li}.down", seed, dtype, device)
                layer.mlp.down_w.copy_(down[:, rank * i_loc:(rank + 1) * i_loc])
            layer.input_norm.fill_(1.0)
            layer.post_attn_norm.fill_(1.0)
            continue
        q_full = _gen((spec.num_heads * d, spec.hidden_size), f"{li}.q", seed, dtype, device)
        k_full = _gen((spec.num_kv_heads * d, spec.hidden_size), f"{li}.k", seed, dtype, device)
        v_full = _gen((spec.num_kv_heads * d, spec.hidden_size), f"{li}.v", seed, dtype, device)
        layer.attn.qkv_w.copy_(torch.cat([
            q_full[rank * hq * d:(rank + 1) * hq * d],
            _kv_slice(k_full, rank, tp, spec.num_kv_heads, d),
            _kv_slice(v_full, rank, tp, spec.num_kv_heads, d),
        ]))
        if layer.attn.qkv_b is not None:
            qb = _gen((spec.num_heads * d,), f"{li}.qb", seed, dtype, device)
            kb = _gen((spec.num_kv_heads * d,), f"{li}.kb", seed, dtype, device)
            vb = _gen((spec.num_kv_heads * d,), f"{li}.vb", seed, dtype, device)
            layer.attn.qkv_b.copy_(torch.cat([
                qb[rank * hq * d:(rank + 1) * hq * d],
                _kv_slice(kb.unsqueeze(1), rank, tp, spec.num_kv_heads,
                          d).squeeze(1),
                _kv_slice(vb.unsqueeze(1), rank, tp, spec.num_kv_heads,
                          d).squeeze(1),
            ]))
        o_full = _gen((spec.hidden_size, spec.num_heads * d), f"{li}.o", seed, dtype, device)
        layer.attn.o_w.copy_(o_full[:, rank * hq * d:(rank + 1) * hq * d])
        if layer.attn.o_b is not None:
            layer.attn.o_b.copy_(_gen((spec.hidden_size,), f"{li}.ob", seed,
                                      dtype, device))
        if layer.attn.sinks is not None:
            sk = _gen((spec.num_heads,), f"{li}.sinks", seed,
                      torch.float32, device, std=0.5)
            layer.attn.sinks.copy_(sk[rank * hq:(rank + 1) * hq])
        if spec.qk_norm:
            layer.attn.q_norm.fill_(1.0)
            layer.attn.k_norm.fill_(1.0)
        if hasattr(layer.mlp, "router_w"):
            _random_init_moe_layer(layer, spec, li, seed, dtype, device, tp, rank)
        else:
            up = _gen((spec.intermediate_size, spec.hidden_size), f"{li}.up", seed, dtype, device)
            if getattr(layer.mlp, "no_gate", False):
                layer.mlp.gate_up_w.copy_(up[rank * i_loc:(rank + 1) * i_loc])
            else:
                gate = _gen((spec.intermediate_size, spec.hidden_size), f"{li}.gate", seed, dtype, device)
                layer.mlp.gate_up_w.copy_(torch.cat([
                    gate[rank * i_loc:(rank + 1) * i_loc],
                    up[rank * i_loc:(rank + 1) * i_loc],
                ]))
            down = _gen((spec.hidden_size, spec.intermediate_size), f"{li}.down", seed, dtype, device)
            layer.mlp.down_w.copy_(down[:, rank * i_loc:(rank + 1) * i_loc])
        layer.input_norm.fill_(1.0)
        layer.post_attn_norm.fill_(1.0)
        if layer.pre_ff_norm is not None:  # Gemma-2 sandwich norms
            layer.pre_ff_norm.fill_(1.0)
            layer.post_ff_norm.fill_(1.0)


def _random_init_mla_attn(layer, spec, li, seed, dtype, device, tp, rank):
    """DeepSeek MLA attention params (models/llama.py MLAAttention):
    q_b / kv_b / o shard by head, the latent projections replicate."""
    a = layer.attn
    nh_loc = spec.num_heads // tp
    dq = spec.qk_nope_head_dim + spec.qk_rope_head_dim
    if spec.q_lora_rank:
        a.q_a_w.copy_(_gen((spec.q_lora_rank, spec.hidden_size),
                           f"{li}.qa", seed, dtype, device))
        a.q_a_norm.fill_(1.0)
        qb = _gen((spec.num_heads * dq, spec.q_lora_rank), f"{li}.qbw",
                  seed, dtype, device)
        a.q_b_w.copy_(qb[rank * nh_loc * dq:(rank + 1) * nh_loc * dq])
    else:
        qw = _gen((spec.num_heads * dq, spec.hidden_size), f"{li}.qw",
                  seed, dtype, device)
        a.q_w.copy_(qw[rank * nh_loc * dq:(rank + 1) * nh_loc * dq])
    a.kv_a_w.copy_(_gen((spec.kv_lora_rank + spec.qk_rope_head_dim,
                         spec.hidden_size), f"{li}.kva", seed, dtype, device))
    a.kv_a_norm.fill_(1.0)
    dkv = spec.qk_nope_head_dim + spec.v_head_dim
    kvb = _gen((spec.num_heads * dkv, spec.kv_lora_rank), f"{li}.kvb",
               seed, dtype, device)
    a.kv_b_w.copy_(kvb[rank * nh_loc * dkv:(rank + 1) * nh_loc * dkv])
    o = _gen((spec.hidden_size, spec.num_heads * spec.v_head_dim),
             f"{li}.o", seed, dtype, device)
    dv = spec.v_head_dim
    a.o_w.copy_(o[:, rank * nh_loc * dv:(rank + 1) * nh_loc * dv])


def _random_init_moe_layer(layer, spec, li, seed, dtype, device, tp, rank):
    mi_loc = spec.moe_intermediate_size // tp
    layer.mlp.router_w.copy_(_gen((spec.num_experts, spec.hidden_size),
                                  f"{li}.router", seed, dtype, device))
    if getattr(layer.mlp, "shared_gate_w", None) is not None:
        layer.mlp.shared_gate_w.copy_(_gen((1, spec.hidden_size),
                                           f"{li}.sgate", seed, dtype,
                                           device))
    if layer.mlp.router_bias is not None:
        layer.mlp.router_bias.copy_(_gen((spec.num_experts,), f"{li}.rbias",
                                         seed, torch.float32, device,
                                         std=0.05))
    if getattr(layer.mlp, "gate_up_b", None) is not None:
        mi = spec.moe_intermediate_size
        gb = _gen((spec.num_experts, mi), f"{li}.egb", seed, dtype, device)
        ub = _gen((spec.num_experts, mi), f"{li}.eub", seed, dtype, device)
        layer.mlp.gate_up_b.copy_(torch.cat([
            gb[:, rank * mi_loc:(rank + 1) * mi_loc],
            ub[:, rank * mi_loc:(rank + 1) * mi_loc]], dim=1))
        layer.mlp.down_b.copy_(_gen((spec.num_experts, spec.hidden_size),
                                    f"{li}.edb", seed, dtype, device))
    if layer.mlp.shared_gate_up_w is not None:
        si_full = spec.moe_intermediate_size * spec.n_shared_experts
        si = si_full // tp
        sg = _gen((si_full, spec.hidden_size), f"{li}.sgate", seed, dtype, device)
        su = _gen((si_full, spec.hidden_size), f"{li}.sup", seed, dtype, device)
        layer.mlp.shared_gate_up_w.copy_(torch.cat([
            sg[rank * si:(rank + 1) * si], su[rank * si:(rank + 1) * si]]))
        sd = _gen((spec.hidden_size, si_full), f"{li}.sdown", seed, dtype, device)
        layer.mlp.shared_down_w.copy_(sd[:, rank * si:(rank + 1) * si])
    for e in range(spec.num_experts):
        gate = _gen((spec.moe_intermediate_size, spec.hidden_size),
                    f"{li}.e{e}.gate", seed, dtype, device)
        up = _gen((spec.moe_intermediate_size, spec.hidden_size),
                  f"{li}.e{e}.up", seed, dtype, device)
        layer.mlp.gate_up_w[e].copy_(torch.cat([
            gate[rank * mi_loc:(rank + 1) * mi_loc],
            up[rank * mi_loc:(rank + 1) * mi_loc]]))
        down = _gen((spec.hidden_size, spec.moe_intermediate_size),
                    f"{li}.e{e}.down", seed, dtype, device)
        layer.mlp.down_w[e].copy_(down[:, rank * mi_loc:(rank + 1) * mi_loc])


def load_safetensors(model, cfg: EngineConfig, model_dir: str | Path) -> None:
    """Load an HF Llama/Qwen checkpoint into the fused layout, sharded for TP."""
    from safetensors import safe_open

    spec = model.spec
    tp, rank = cfg.tp_size, cfg.tp_rank
    d = spec.head_dim
    hq, hkv = spec.num_heads // tp, max(1, spec.num_kv_heads // tp)
    i_loc = spec.intermediate_size // tp
    model_dir = Path(model_dir)
    files = sorted(model_dir.glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no safetensors under {model_dir}")

    def row_shard(t, n):  # column-parallel: shard output rows
        per = t.shape[0] // tp if n is None else n
        return t[rank * per:(rank + 1) * per]

    tensors: dict[str, torch.Tensor] = {}
    for f in files:
        with safe_open(str(f), framework="pt") as sf:
            for name in sf.keys():
                tensors[name] = sf.get_tensor(name)

    from .quantized import maybe_dequant, quant_config

    qc = quant_config(model_dir)  # GPTQ/AWQ: dequant packed linears on load

    def get(name):
        t = maybe_dequant(tensors, name, qc)
        return t.to(model.dtype)

    pre = "model."
    off = getattr(model, "layer_offset", 0)
    if model.embed is not None:
        model.embed.copy_(get(pre + "embed_tokens.weight"))
    if model.final_norm is not None:
        fn = get(pre + "norm.weight")
        model.final_norm.copy_(fn + 1 if spec.sandwich_norms else fn)
    if model.lm_head is not None and not spec.tie_word_embeddings:
        model.lm_head.copy_(get("lm_head.weight"))
    for local_i, layer in enumerate(model.layers):
        li = off + local_i
        p = f"{pre}layers.{li}."
        if p + "self_attn.qkv_proj.weight" in tensors:
            # Phi-3/4: fused qkv checkpoint tensor [nq + 2*nk, h] — split
            # then shard exactly like separate projections
            fused = get(p + "self_attn.qkv_proj.weight")
            nq_f = spec.num_heads * d
            nk_f = spec.num_kv_heads * d
            tensors[p + "self_attn.q_proj.weight"] = fused[:nq_f]
            tensors[p + "self_attn.k_proj.weight"] = fused[nq_f:nq_f + nk_f]
            tensors[p + "self_attn.v_proj.weight"] = fused[nq_f + nk_f:]
        if p + "mlp.gate_up_proj.weight" in tensors \
                and not hasattr(layer.mlp, "router_w"):
            fused = get(p + "mlp.gate_up_proj.weight")
            ii = spec.intermediate_size
            tensors[p + "mlp.gate_proj.weight"] = fused[:ii]
            tensors[p + "mlp.up_proj.weight"] = fused[ii:]
        if spec.kv_lora_rank:
            # DeepSeek MLA: latent projections replicate, per-head
            # projections shard by head (models/llama.py MLAAttention)
            a = layer.attn
            nh_loc = spec.num_heads // tp
            dq = spec.qk_nope_head_dim + spec.qk_rope_head_dim
            if spec.q_lora_rank:
                a.q_a_w.copy_(get(p + "self_attn.q_a_proj.weight"))
                a.q_a_norm.copy_(get(p + "self_attn.q_a_layernorm.weight"))
                a.q_b_w.copy_(row_shard(
                    get(p + "self_attn.q_b_proj.weight"), nh_loc * dq))
            else:
                a.q_w.copy_(row_shard(
                    get(p + "self_attn.q_proj.weight"), nh_loc * dq))
            a.kv_a_w.copy_(get(p + "self_attn.kv_a_proj_with_mqa.weight"))
            a.kv_a_norm.copy_(get(p + "self_attn.kv_a_layernorm.weight"))
            dkv = spec.qk_nope_head_dim + spec.v_head_dim
            a.kv_b_w.copy_(row_shard(
                get(p + "self_attn.kv_b_proj.weight"), nh_loc * dkv))
            o = get(p + "self_attn.o_proj.weight")
            dv = spec.v_head_dim
            a.o_w.copy_(
                o[:, rank * nh_loc * dv:(rank + 1) * nh_loc * dv])
        else:
            q = row_shard(get(p + "self_attn.q_proj.weight"), hq * d)
            k = _kv_slice(get(p + "self_attn.k_proj.weight"), rank, tp,
                          spec.num_kv_heads, d)
            v = _kv_slice(get(p + "self_attn.v_proj.weight"), rank, tp,
                          spec.num_kv_heads, d)
            layer.attn.qkv_w.copy_(torch.cat([q, k, v]))
            if layer.attn.qkv_b is not None:
                layer.attn.qkv_b.copy_(torch.cat([
                    row_shard(get(p + "self_attn.q_proj.bias"), hq * d),
                    _kv_slice(get(p + "self_attn.k_proj.bias").unsqueeze(1),
                              rank, tp, spec.num_kv_heads, d).squeeze(1),
                    _kv_slice(get(p + "self_attn.v_proj.bias").unsqueeze(1),
                              rank, tp, spec.num_kv_heads, d).squeeze(1),
                ]))
            o = get(p + "self_attn.o_proj.weight")
            layer.attn.o_w.copy_(o[:, rank * hq * d:(rank + 1) * hq * d])
            if layer.attn.o_b is not None                 and p + "self_attn.o_proj.bias" in tensors:
                layer.attn.o_b.copy_(get(p + "self_attn.o_proj.bias"))
            if layer.attn.sinks is not None and p + "self_attn.sinks" in tensors:
                sk = tensors[p + "self_attn.sinks"].float()
                layer.attn.sinks.copy_(sk[rank * hq:(rank + 1) * hq])
            if spec.qk_norm:
                off = 1 if spec.sandwich_norms else 0  # Gemma: (1+w) norms
                qn_name = (p + "self_attn.q_norm.weight"
                           if p + "self_attn.q_norm.weight" in tensors
                           else p + "self_attn.query_layernorm.weight")
                kn_name = (p + "self_attn.k_norm.weight"
                           if p + "self_attn.k_norm.weight" in tensors
                           else p + "self_attn.key_layernorm.weight")
                qn = get(qn_name) + off
                kn = get(kn_name) + off
                if spec.qk_norm_full:  # OLMo-2: weights shard with heads
                    qn = qn[rank * hq * d:(rank + 1) * hq * d]
                    kn = _kv_slice(kn.unsqueeze(1), rank, tp,
                                   spec.num_kv_heads, d).squeeze(1)
                layer.attn.q_norm.copy_(qn)
                layer.attn.k_norm.copy_(kn)
        if hasattr(layer.mlp, "router_w")                 and spec.architecture.startswith("GptOss"):
            # GPT-OSS: router `mlp.router.{weight,bias}`; experts stored
            # TRANSPOSED ([E, h, 2i] / [E, i, h]) with INTERLEAVED
            # gate/up columns ([..., ::2] gate) + per-expert biases —
            # de-interleave into our fused [gate; up] layout at load
            layer.mlp.router_w.copy_(get(p + "mlp.router.weight"))
            layer.mlp.router_bias.copy_(tensors[p + "mlp.router.bias"].float())
            mi_loc = spec.moe_intermediate_size // tp
            gu = get(p + "mlp.experts.gate_up_proj")     # [E, h, 2i]
            gate = gu[:, :, 0::2].transpose(1, 2)        # [E, i, h]
            up = gu[:, :, 1::2].transpose(1, 2)
            layer.mlp.gate_up_w.copy_(torch.cat([
                gate[:, rank * mi_loc:(rank + 1) * mi_loc],
                up[:, rank * mi_loc:(rank + 1) * mi_loc]], dim=1))
            gub = get(p + "mlp.experts.gate_up_proj_bias")  # [E, 2i]
            layer.mlp.gate_up_b.copy_(torch.cat([
                gub[:, 0::2][:, rank * mi_loc:(rank + 1) * mi_loc],
                gub[:, 1::2][:, rank * mi_loc:(rank + 1) * mi_loc]], dim=1))
            dn = get(p + "mlp.experts.down_proj")        # [E, i, h]
            layer.mlp.down_w.copy_(
                dn.transpose(1, 2)[:, :, rank * mi_loc:(rank + 1) * mi_loc])
            layer.mlp.down_b.copy_(get(p + "mlp.experts.down_proj_bias"))
            layer.input_norm.copy_(get(p + "input_layernorm.weight"))
            layer.post_attn_norm.copy_(
                get(p + "post_attention_layernorm.weight"))
            continue
        if hasattr(layer.mlp, "router_w"):
            # MoE (Qwen3-MoE `mlp.gate`, Mixtral `block_sparse_moe.gate`)
            router = (p + "mlp.gate.weight" if p + "mlp.gate.weight" in tensors
                      else p + "block_sparse_moe.gate.weight")
            layer.mlp.router_w.copy_(get(router))
            if layer.mlp.router_bias is not None:
                # DeepSeek keeps it on the gate; MiniMax-M2 on the block
                for bn in ("mlp.gate.e_score_correction_bias",
                           "mlp.e_score_correction_bias"):
                    if p + bn in tensors:
                        layer.mlp.router_bias.copy_(tensors[p + bn].float())
                        break
            if getattr(layer.mlp, "shared_gate_w", None) is not None \
                    and p + "mlp.shared_expert_gate.weight" in tensors:
                layer.mlp.shared_gate_w.copy_(
                    get(p + "mlp.shared_expert_gate.weight"))
            if layer.mlp.shared_gate_up_w is not None:
                si = layer.mlp.shared_i
                # DeepSeek/GLM name it shared_expertS, Qwen2-MoE singular
                sp = (p + "mlp.shared_experts."
                      if p + "mlp.shared_experts.gate_proj.weight"
                      in tensors else p + "mlp.shared_expert.")
                sg = get(sp + "gate_proj.weight")
                su = get(sp + "up_proj.weight")
                layer.mlp.shared_gate_up_w.copy_(torch.cat([
                    sg[rank * si:(rank + 1) * si],
                    su[rank * si:(rank + 1) * si]]))
                sd = get(sp + "down_proj.weight")
                layer.mlp.shared_down_w.copy_(
                    sd[:, rank * si:(rank + 1) * si])
            mi_loc = spec.moe_intermediate_size // tp
            mi = spec.moe_intermediate_size
            if p + "mlp.experts.gate_up_proj" in tensors:
                # transformers >=5 fused layout [E, 2mi, h] / [E, h, mi]
                gu = get(p + "mlp.experts.gate_up_proj")
                dn = get(p + "mlp.experts.down_proj")
                layer.mlp.gate_up_w.copy_(torch.cat([
                    gu[:, rank * mi_loc:(rank + 1) * mi_loc],
                    gu[:, mi + rank * mi_loc:mi + (rank + 1) * mi_loc],
                ], dim=1))
                layer.mlp.down_w.copy_(
                    dn[:, :, rank * mi_loc:(rank + 1) * mi_loc])
                layer.input_norm.copy_(get(p + "input_layernorm.weight"))
                layer.post_attn_norm.copy_(
                    get(p + "post_attention_layernorm.weight"))
                continue
            for e in range(spec.num_experts):
                if f"{p}mlp.experts.{e}.gate_proj.weight" in tensors:
                    ep = f"{p}mlp.experts.{e}."
                    gname, uname, dname = "gate_proj", "up_proj", "down_proj"
                else:  # mixtral naming: w1=gate, w3=up, w2=down
                    ep = f"{p}block_sparse_moe.experts.{e}."
                    gname, uname, dname = "w1", "w3", "w2"
                layer.mlp.gate_up_w[e].copy_(torch.cat([
                    row_shard(get(ep + gname + ".weight"), mi_loc),
                    row_shard(get(ep + uname + ".weight"), mi_loc),
                ]))
                dn = get(ep + dname + ".weight")
                layer.mlp.down_w[e].copy_(dn[:, rank * mi_loc:(rank + 1) * mi_loc])
        elif getattr(layer.mlp, "no_gate", False):
            # Arcee: up_proj only
            layer.mlp.gate_up_w.copy_(
                row_shard(get(p + "mlp.up_proj.weight"), i_loc))
            dn = get(p + "mlp.down_proj.weight")
            layer.mlp.down_w.copy_(dn[:, rank * i_loc:(rank + 1) * i_loc])
        else:
            layer.mlp.gate_up_w.copy_(torch.cat([
                row_shard(get(p + "mlp.gate_proj.weight"), i_loc),
                row_shard(get(p + "mlp.up_proj.weight"), i_loc),
            ]))
            dn = get(p + "mlp.down_proj.weight")
            layer.mlp.down_w.copy_(dn[:, rank * i_loc:(rank + 1) * i_loc])
        if spec.parallel_block:
            # Cohere: ONE shared input LayerNorm per layer; the
            # post_attn_norm slot is unused (left at identity)
            layer.input_norm.copy_(get(p + "input_layernorm.weight"))
            layer.post_attn_norm.fill_(1.0)
        elif spec.norm_after:
            # OLMo-2: no input norms — our input_norm slot holds the
            # post-attention norm, post_attn_norm the post-ffn one
            layer.input_norm.copy_(
                get(p + "post_attention_layernorm.weight"))
            layer.post_attn_norm.copy_(
                get(p + "post_feedforward_layernorm.weight"))
        elif spec.sandwich_norms \
                and p + "post_self_attn_layernorm.weight" in tensors:
            # GLM-4 dense: same four-norm sandwich flow under different
            # HF names, plain RMSNorm weights (no offset)
            layer.input_norm.copy_(get(p + "input_layernorm.weight"))
            layer.post_attn_norm.copy_(
                get(p + "post_self_attn_layernorm.weight"))
            layer.pre_ff_norm.copy_(
                get(p + "post_attention_layernorm.weight"))
            layer.post_ff_norm.copy_(
                get(p + "post_mlp_layernorm.weight"))
        elif spec.sandwich_norms:
            # Gemma-2 RMSNorm multiplies by (1 + w); store the EFFECTIVE
            # weight so the shared rms_norm kernel applies
            layer.input_norm.copy_(get(p + "input_layernorm.weight") + 1)
            layer.post_attn_norm.copy_(
                get(p + "post_attention_layernorm.weight") + 1)
            layer.pre_ff_norm.copy_(
                get(p + "pre_feedforward_layernorm.weight") + 1)
            layer.post_ff_norm.copy_(
                get(p + "post_feedforward_layernorm.weight") + 1)
        else:
            layer.input_norm.copy_(get(p + "input_layernorm.weight"))
            layer.post_attn_norm.copy_(
                get(p + "post_attention_layernorm.weight"))


def merge_lora(model, cfg: EngineConfig, adapter_dir: str | Path) -> int:
    """Merge a PEFT LoRA adapter into the serving weights at load time
    (reference parity: Model.lora_list mounting, ref schemas/models.py:468;
    merged adapters add zero runtime cost — dynamic multi-adapter serving
    is a later round).

    Supports q/k/v/o/gate/up/down projection adapters in the standard PEFT
    naming (`base_model.model.model.layers.N.self_attn.q_proj.lora_A.weight`
    etc). Returns the number of merged tensors.
    """
    import json as _json

    from safetensors import safe_open

    adapter_dir = Path(adapter_dir)
    cfg_path = adapter_dir / "adapter_config.json"
    alpha, r = 16.0, 8.0
    if cfg_path.exists():
        with open(cfg_path) as f:
            ac = _json.load(f)
        alpha = float(ac.get("lora_alpha", alpha))
        r = float(ac.get("r", r))
    scaling = alpha / r

    files = sorted(adapter_dir.glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no adapter safetensors under {adapter_dir}")
    tensors: dict[str, torch.Tensor] = {}
    for f in files:
        with safe_open(str(f), framework="pt") as sf:
            for name in sf.keys():
                tensors[name] = sf.get_tensor(name)

    spec = model.spec
    tp, rank = cfg.tp_size, cfg.tp_rank
    d = spec.head_dim
    hq, hkv = spec.num_heads // tp, max(1, spec.num_kv_heads // tp)
    i_loc = spec.intermediate_size // tp

    def find(li: int, proj: str, which: str) -> torch.Tensor | None:
        for key, t in tensors.items():
            if f"layers.{li}." in key and f"{proj}." in key and f"lora_{which}" in key:
                return t.float()
        return None

    merged = 0
    off = getattr(model, "layer_offset", 0)
    for local_i, layer in enumerate(model.layers):
        li = off + local_i
        # fused qkv: rows [0,hq*d) = q shard, then k, then v
        offsets = {
            "q_proj": (0, hq * d, rank * hq * d),
            "k_proj": (hq * d, hkv * d, rank * hkv * d),
            "v_proj": ((hq + hkv) * d, hkv * d, rank * hkv * d),
        }
        for proj, (dst_off, rows, src_off) in offsets.items():
            A, B = find(li, proj, "A"), find(li, proj, "B")
            if A is None or B is None:
                continue
            delta = (B @ A) * scaling              # [out_full, in]
            shard = delta[src_off:src_off + rows]
            w = layer.attn.qkv_w.data
            w[dst_off:dst_off + rows] += shard.to(w.dtype).to(w.device)
            merged += 1
        A, B = find(li, "o_proj", "A"), find(li, "o_proj", "B")
        if A is not None and B is not None:
            delta = (B @ A) * scaling              # [h, hq_full*d]
            shard = delta[:, rank * hq * d:(rank + 1) * hq * d]
            w = layer.attn.o_w.data
            w += shard.to(w.dtype).to(w.device)
            merged += 1
        for proj, dst_off, rows, src_off in (
            ("gate_proj", 0, i_loc, rank * i_loc),
            ("up_proj", i_loc, i_loc, rank * i_loc),
        ):
            A, B = find(li, proj, "A"), find(li, proj, "B")
            if A is None or B is None:
                continue
            delta = (B @ A) * scaling
            shard = delta[src_off:src_off + rows]
            w = layer.mlp.gate_up_w.data
            w[dst_off:dst_off + rows] += shard.to(w.dtype).to(w.device)
            merged += 1
        A, B = find(li, "down_proj", "A"), find(li, "down_proj", "B")
        if A is not None and B is not None:
            delta = (B @ A) * scaling              # [h, i_full]
            shard = delta[:, rank * i_loc:(rank + 1) * i_loc]
            w = layer.mlp.down_w.data
            w += shard.to(w.dtype).to(w.device)
            merged += 1
    return merged

def load_gguf(model, cfg: EngineConfig, gguf_path: str | Path) -> None:
    """Load a GGUF checkpoint (llama.cpp naming: blk.N.attn_q / ffn_gate /
    token_embd / output ...) into the fused bf16 serving layout, dequantizing
    Q8_0/Q4_0/Q4_1/Q4_K/Q6_K blocks at load time (utils/gguf.py).

    Reference parity: the reference schedules GGUF models but delegates
    execution to llama-box containers; here the first-
    party engine executes them on the same MFMA bf16 path as safetensors.

    llama-arch GGUFs store q/k projections PERMUTED for ggml's interleaved
    rope (convert_hf_to_gguf.py LlamaModel.permute); our rope kernel uses
    the HF half-split convention, so those rows are un-permuted here.
    qwen2/qwen3-arch GGUFs use neox rope and need no permute.
    """
    from ..utils.gguf import read_gguf, read_tensor

    spec = model.spec
    tp, rank = cfg.tp_size, cfg.tp_rank
    d = spec.head_dim
    hq, hkv = spec.num_heads // tp, max(1, spec.num_kv_heads // tp)
    i_loc = spec.intermediate_size // tp
    info = read_gguf(gguf_path)
    by_name = {t.name: t for t in info.tensors}
    permuted_qk = info.architecture == "llama"

    def get(name, unpermute_heads: int | None = None):
        t = by_name[name]
        w = read_tensor(gguf_path, info, t)
        if unpermute_heads:
            # inverse of convert_hf_to_gguf permute:
            #   permute  = reshape(h, 2, d/2, in).swapaxes(1, 2)
            #   inverse  = reshape(h, d/2, 2, in).swapaxes(1, 2)
            nh = unpermute_heads
            w = w.reshape(nh, d // 2, 2, *w.shape[1:]).transpose(1, 2) \
                 .reshape(nh * d, *w.shape[1:])
        return w.to(model.dtype)

    def row_shard(t, per):
        return t[rank * per:(rank + 1) * per]

    off = getattr(model, "layer_offset", 0)
    if model.embed is not None:
        model.embed.copy_(get("token_embd.weight"))
    if model.final_norm is not None:
        model.final_norm.copy_(get("output_norm.weight"))
    if model.lm_head is not None and not spec.tie_word_embeddings:
        name = "output.weight" if "output.weight" in by_name else "token_embd.weight"
        model.lm_head.copy_(get(name))
    up_q = spec.num_heads if permuted_qk else None
    up_k = spec.num_kv_heads if permuted_qk else None
    for local_i, layer in enumerate(model.layers):
        li = off + local_i
        p = f"blk.{li}."
        q = row_shard(get(p + "attn_q.weight", up_q), hq * d)
        k = row_shard(get(p + "attn_k.weight", up_k), hkv * d)
        v = row_shard(get(p + "attn_v.weight"), hkv * d)
        layer.attn.qkv_w.copy_(torch.cat([q, k, v]))
        if layer.attn.qkv_b is not None:
            layer.attn.qkv_b.copy_(torch.cat([
                row_shard(get(p + "attn_q.bias", up_q), hq * d),
                row_shard(get(p + "attn_k.bias", up_k), hkv * d),
                row_shard(get(p + "attn_v.bias"), hkv * d),
            ]))
        o = get(p + "attn_output.weight")
        layer.attn.o_w.copy_(o[:, rank * hq * d:(rank + 1) * hq * d])
        if spec.qk_norm:
            layer.attn.q_norm.copy_(get(p + "attn_q_norm.weight"))
            layer.attn.k_norm.copy_(get(p + "attn_k_norm.weight"))
        layer.mlp.gate_up_w.copy_(torch.cat([
            row_shard(get(p + "ffn_gate.weight"), i_loc),
            row_shard(get(p + "ffn_up.weight"), i_loc),
        ]))
        dn = get(p + "ffn_down.weight")
        layer.mlp.down_w.copy_(dn[:, rank * i_loc:(rank + 1) * i_loc])
        layer.input_norm.copy_(get(p + "attn_norm.weight"))
        layer.post_attn_norm.copy_(get(p + "ffn_norm.weight"))


def convert_to_w4_runtime(model, cfg: EngineConfig) -> int:
    """Quantize the big serving weights to the packed W4 runtime format
    (models/quantized.py) and free their bf16 parameters: weights stay
    resident int4 in HBM and decode GEMMs run the in-register dequant
    kernel (ops/csrc/w4_gemm.hip). Returns the number of packed tensors.

    Uniform path for every weight source: random-init, safetensors, and
    dequantized GPTQ/AWQ/GGUF checkpoints all quantize post-load
    (asymmetric per-128 groups). Ineligible shapes (N%64 / K%128) keep
    bf16 — mixed execution is fine. Reference capability: vLLM
    --quantization gptq/awq serving, re-designed as a
    CDNA4 fragment-ordered format.
    """
    from .llama import W4Pack
    from .quantized import pack_w4_runtime

    def quantize(w: torch.Tensor):
        N, K = w.shape
        if N % 64 or K % 128:
            return None
        wf = w.float().view(N, K // 128, 128)
        mn = wf.amin(-1)
        mx = wf.amax(-1)
        scale = ((mx - mn) / 15).clamp_min(1e-8)
        zero = (-mn / scale).round().clamp(0, 15)
        q = ((wf / scale.unsqueeze(-1)) + zero.unsqueeze(-1)) \
            .round().clamp(0, 15).view(N, K).to(torch.uint8)
        return pack_w4_runtime(q, scale, zero, 128)

    n_packed = 0

    def pack_site(mod, wname: str, pname: str) -> None:
        nonlocal n_packed
        w = getattr(mod, wname, None)  # MLA attention has no fused qkv
        if w is None or w.numel() == 0:
            return
        pk = quantize(w.data)
        if pk is None:
            return
        setattr(mod, pname, W4Pack(*pk))
        w.data = torch.empty(0, dtype=w.dtype, device=w.device)
        n_packed += 1

    for layer in model.layers:
        if model.spec.kv_lora_rank:
            # MLA attention routes its projections through plain
            # F.linear/einsums (no qlinear dispatch yet) — leave them
            # bf16; W4 there is an r3 follow-up with the MLA kernels
            pass
        else:
            pack_site(layer.attn, "qkv_w", "qkv_pack")
            pack_site(layer.attn, "o_w", "o_pack")
        if model.spec.num_experts == 0:
            pack_site(layer.mlp, "gate_up_w", "gate_up_pack")
            pack_site(layer.mlp, "down_w", "down_pack")
        elif hasattr(layer.mlp, "router_w"):
            # MoE: per-expert packing of the 3-D banks — the capacity
            # bulk of MoE checkpoints lives here (Qwen3-235B: ~97% of
            # weights). Bank consumers dequant transiently per call;
            # the loop path dequants per expert via qlinear.
            for wname, pname in (("gate_up_w", "gate_up_packs"),
                                 ("down_w", "down_packs")):
                bank = getattr(layer.mlp, wname)
                pks = []
                for e in range(bank.shape[0]):
                    pk = quantize(bank.data[e])
                    if pk is None:
                        pks = None
                        break
                    pks.append(W4Pack(*pk))
                if pks is not None:
                    setattr(layer.mlp, pname, pks)
                    bank.data = torch.empty(0, dtype=bank.dtype,
                                            device=bank.device)
                    n_packed += len(pks)
        elif model.spec.num_experts > 0:
            # dense-first layer of an MoE model: plain MLP, pack normally
            pack_site(layer.mlp, "gate_up_w", "gate_up_pack")
            pack_site(layer.mlp, "down_w", "down_pack")
    if model.lm_head is not None and not model.spec.tie_word_embeddings:
        w = model.lm_head
        pk = quantize(w.data)
        if pk is not None:
            from .llama import W4Pack as _P

            model.lm_head_pack = _P(*pk)
            w.data = torch.empty(0, dtype=w.dtype, device=w.device)
            n_packed += 1
    if model.device.type == "cuda":
        torch.cuda.empty_cache()
    return n_packed
