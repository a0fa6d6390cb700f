"""Engine + model configuration for the first-party CDNA4 inference engine.

ModelSpec mirrors the fields the reference reads out of HF config.json via
AutoConfig (reference: gpustack/scheduler/scheduler.py:216 evaluation path);
we parse config.json directly so random-init serving needs no checkpoint.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from pathlib import Path


@dataclass
class ModelSpec:
    architecture: str = "LlamaForCausalLM"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    rope_scaling: dict | None = None
    rms_norm_eps: float = 1e-5
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    attention_bias: bool = False  # qwen2-style qkv bias
    qk_norm: bool = False         # qwen3-style per-head q/k RMSNorm
    eos_token_id: int = 128001
    # MoE (Qwen3-MoE / Mixtral): num_experts > 0 makes every decoder
    # layer's MLP a router + expert bank (models/llama.py MoEMLP)
    num_experts: int = 0
    num_experts_per_tok: int = 0
    moe_intermediate_size: int = 0
    norm_topk_prob: bool = True
    # GLM-4.5 / DeepSeek-style MoE extensions (Glm4MoeForCausalLM):
    # sigmoid router scores + learned correction bias + optional grouped
    # top-k; a SHARED dense expert added to every token; the first k
    # layers use a plain dense MLP
    router_mode: str = "softmax"          # "softmax" | "sigmoid_bias"
    n_shared_experts: int = 0
    shared_expert_gated: bool = False    # Qwen2-MoE: sigmoid(gate(x)) scales
                                         # the shared expert's output
    first_k_dense_replace: int = 0
    routed_scaling_factor: float = 1.0
    n_group: int = 1
    topk_group: int = 1
    # rotary applied to the first head_dim*partial_rotary_factor dims only
    partial_rotary_factor: float = 1.0
    # GPT-OSS extensions (GptOssForCausalLM):
    # - attention SINKS: per-head learned logit joining the softmax
    #   denominator only
    # - alternating sliding-window / full attention layers (layer_types)
    # - clamped-swiglu expert activation + expert/router biases
    # - bias on o_proj (attention_bias covers q/k/v)
    attention_sinks: bool = False
    sliding_window: int = 0
    layer_types: tuple | None = None     # ("sliding_attention"|"full_attention", ...)
    moe_act: str = "silu"                # "silu" | "clamped_swiglu"
    moe_bias: bool = False               # expert gate_up/down biases
    router_logit_bias: bool = False      # bias added to router logits
    o_proj_bias: bool = False
    # DeepSeek V2/V3/R1 MLA (DeepseekV3ForCausalLM, models/llama.py
    # MLAAttention): kv_lora_rank > 0 switches attention to the latent
    # formulation — the paged cache stores [kv_lora_rank + qk_rope]
    # per token per layer instead of 2 * kv_heads * head_dim
    q_lora_rank: int = 0
    kv_lora_rank: int = 0
    qk_nope_head_dim: int = 0
    qk_rope_head_dim: int = 0
    v_head_dim: int = 0
    rope_interleave: bool = True         # deepseek weights: paired dims
    # Gemma-2 (Gemma2ForCausalLM): sandwich layer norms (post-attn and
    # post-ffn norms applied BEFORE the residual adds), sqrt(h)-scaled
    # embeddings, tanh logit softcapping in attention and on the final
    # logits, custom attention scale, GeGLU MLP; sliding window
    # alternates via layer_types (even layers) like GPT-OSS
    sandwich_norms: bool = False
    # OLMo-2 (Olmo2ForCausalLM): NO input norms — RMSNorm applied to each
    # sublayer's OUTPUT before the residual add; qk-norm runs over the
    # FULL projected q/k vectors (num_heads*head_dim), not per head
    norm_after: bool = False
    qk_norm_full: bool = False
    # Granite (GraniteForCausalLM): llama graph + scalar multipliers —
    # embeddings (embed_scale above), attention scale (attn_scale above),
    # sublayer outputs before the residual adds, and a final logits
    # DIVISOR
    residual_multiplier: float = 0.0     # 0 = off (1.0)
    logits_scaling: float = 0.0          # 0 = off (divide logits by this)
    # rotary application layout: "neox" rotate-half (llama lineage) or
    # "pairwise" GPT-J-style adjacent-pair rotation (Ernie-4.5; Hunyuan
    # and MiniMax share it — SURVEY family matrix)
    rope_mode: str = "neox"
    # Hunyuan-dense: per-head qk-norm applied AFTER rope (Qwen3 norms
    # before rope)
    qk_norm_after_rope: bool = False
    # SmolLM3 (SmolLM3ForCausalLM): per-layer rope switch — entry 0 means
    # a NoPE layer (no rotary at all); None = rope everywhere
    no_rope_layers: tuple | None = None
    # Gemma-3: sliding layers rope at a LOCAL base frequency (10k) while
    # full-attention layers use rope_theta (1M, linearly scaled) — the
    # model holds two cos/sin caches and each layer picks by window
    rope_local_theta: float = 0.0
    embed_scale: float = 0.0             # 0 = no scaling
    attn_logit_softcap: float = 0.0      # 0 = off
    final_logit_softcap: float = 0.0
    attn_scale: float = 0.0              # 0 = 1/sqrt(head_dim)
    mlp_act: str = "silu"                # "silu" | "gelu_tanh" | "relu2"
    norm_type: str = "rmsnorm"           # "layernorm": Cohere mean-centered
    parallel_block: bool = False         # Cohere: x + attn(ln(x)) + mlp(ln(x))
    logits_multiplier: float = 0.0       # Cohere logit_scale (multiplies)
    mlp_no_gate: bool = False            # Arcee: down(act(up(x))), no gate

    @property
    def gqa_ratio(self) -> int:
        return self.num_heads // self.num_kv_heads

    def kv_bytes_per_token(self, dtype_size: int = 2) -> int:
        if self.kv_lora_rank:
            # MLA: one compressed latent row per token per layer
            return (self.num_layers
                    * (self.kv_lora_rank + self.qk_rope_head_dim)
                    * dtype_size)
        return 2 * self.num_layers * self.num_kv_heads * self.head_dim * dtype_size

    def weight_bytes(self, dtype_size: int = 2) -> int:
        h, i, v = self.hidden_size, self.intermediate_size, self.vocab_size
        if self.kv_lora_rank:
            dq = self.qk_nope_head_dim + self.qk_rope_head_dim
            qp = (self.q_lora_rank * (h + self.num_heads * dq)
                  if self.q_lora_rank else h * self.num_heads * dq)
            qkv = (qp + h * (self.kv_lora_rank + self.qk_rope_head_dim)
                   + self.kv_lora_rank * self.num_heads
                   * (self.qk_nope_head_dim + self.v_head_dim))
            o = self.num_heads * self.v_head_dim * h
        else:
            qkv = h * (self.num_heads + 2 * self.num_kv_heads) * self.head_dim
            o = self.num_heads * self.head_dim * h
        attn_norm = qkv + o + 2 * h
        dense_mlp = 3 * h * i
        if self.num_experts > 0:
            moe_mlp = (self.num_experts * 3 * h * self.moe_intermediate_size
                       + self.num_experts * h  # router
                       + 3 * h * self.moe_intermediate_size
                       * self.n_shared_experts)
            k_dense = min(self.first_k_dense_replace, self.num_layers)
            total_layers = (attn_norm * self.num_layers
                            + dense_mlp * k_dense
                            + moe_mlp * (self.num_layers - k_dense))
        else:
            total_layers = (attn_norm + dense_mlp) * self.num_layers
        emb = v * h * (1 if self.tie_word_embeddings else 2)
        return (total_layers + emb + h) * dtype_size

    @staticmethod
    def _rope_scaling_from(cfg: dict) -> dict | None:
        sc = cfg.get("rope_scaling") or (
            (cfg.get("rope_parameters") or None)
            if (cfg.get("rope_parameters") or {}).get("rope_type", "default")
            not in ("default",) else None)
        if sc and sc.get("rope_type", sc.get("type")) == "longrope" \
                and "original_max_position_embeddings" not in sc:
            # Phi-3/4 keep the pretraining length top-level
            sc = dict(sc)
            sc["original_max_position_embeddings"] = cfg.get(
                "original_max_position_embeddings",
                cfg.get("max_position_embeddings", 4096))
        return sc

    @classmethod
    def from_hf_config(cls, cfg: dict) -> "ModelSpec":
        arch = (cfg.get("architectures") or ["LlamaForCausalLM"])[0]
        if arch.startswith("Cohere") and cfg.get("use_qk_norm"):
            raise NotImplementedError(
                "Cohere use_qk_norm (per-head LayerNorm, Command-A) is "
                "not implemented yet")
        # OLMo-3 keys rope_parameters BY LAYER TYPE: full_attention
        # defines the main rope (theta + any scaling); sliding layers
        # rope unscaled at their own theta (the Gemma-3 dual-cache path).
        slide_theta = 0.0
        rp = cfg.get("rope_parameters")
        if isinstance(rp, dict) and "full_attention" in rp:
            cfg = dict(cfg)
            cfg["rope_parameters"] = rp.get("full_attention") or {}
            slide_theta = (rp.get("sliding_attention") or {}).get(
                "rope_theta", 0.0) or 0.0
        nh = cfg.get("num_attention_heads", 32)
        hd = cfg.get("head_dim") or cfg.get("hidden_size", 4096) // nh
        eos = cfg.get("eos_token_id", 2)
        if isinstance(eos, list):
            eos = eos[0]
        return cls(
            architecture=arch,
            vocab_size=cfg.get("vocab_size", 32000),
            hidden_size=cfg.get("hidden_size", 4096),
            intermediate_size=cfg.get("intermediate_size", 11008),
            num_layers=cfg.get("num_hidden_layers", 32),
            num_heads=nh,
            num_kv_heads=cfg.get("num_key_value_heads", nh),
            head_dim=hd,
            rope_theta=cfg.get("rope_theta")
            or (cfg.get("rope_parameters") or {}).get("rope_theta", 10000.0),
            rope_scaling=cls._rope_scaling_from(cfg),
            rms_norm_eps=(cfg.get("layer_norm_eps", 1e-5)
                          if arch.startswith("Cohere")
                          else cfg.get("rms_norm_eps", 1e-6)),
            max_position_embeddings=cfg.get("max_position_embeddings", 4096),
            tie_word_embeddings=cfg.get("tie_word_embeddings", False),
            attention_bias=(arch.startswith("Qwen2")
                            or arch.startswith("SeedOss")
                            or (arch in ("GlmForCausalLM",
                                         "Glm4ForCausalLM")
                                and bool(cfg.get("attention_bias", True)))
                            or (arch.startswith("Ernie4_5")
                                and bool(cfg.get("use_bias")))),
            qk_norm=(arch.startswith("Qwen3") or arch.startswith("Gemma3")
                     or arch.startswith(("Olmo2", "Olmo3", "Exaone4"))
                     or arch.startswith(("Olmoe", "MiniMaxM2", "Dots1"))
                     or arch.startswith("HunYuan")
                     or bool(cfg.get("use_qk_norm", False))),
            qk_norm_after_rope=arch.startswith("HunYuan"),
            eos_token_id=eos,
            num_experts=cfg.get("num_experts",
                                cfg.get("num_local_experts",
                                        cfg.get("n_routed_experts", 0))) or 0,
            num_experts_per_tok=cfg.get("num_experts_per_tok", 0) or 0,
            moe_intermediate_size=cfg.get("moe_intermediate_size",
                                          cfg.get("intermediate_size", 0))
            if (cfg.get("num_experts") or cfg.get("num_local_experts")
                or cfg.get("n_routed_experts")) else 0,
            norm_topk_prob=bool(cfg.get("norm_topk_prob", True)),
            router_mode=("sigmoid_bias"
                         if (arch.startswith("Glm4Moe")
                             or arch.startswith(("MiniMaxM2", "Dots1"))
                             or arch.startswith("Deepseek"))
                         else "softmax"),
            n_shared_experts=(cfg.get("n_shared_experts", 0) or 0)
            if not arch.startswith("Qwen2Moe")
            # Qwen2-MoE sizes the shared expert directly; express it as
            # a multiple of the routed intermediate (5632 = 4 x 1408)
            else (cfg.get("shared_expert_intermediate_size", 0)
                  // max(cfg.get("moe_intermediate_size", 1), 1)),
            shared_expert_gated=arch.startswith("Qwen2Moe"),
            first_k_dense_replace=cfg.get("first_k_dense_replace", 0) or 0,
            routed_scaling_factor=cfg.get("routed_scaling_factor", 1.0) or 1.0,
            n_group=cfg.get("n_group", 1) or 1,
            topk_group=cfg.get("topk_group", 1) or 1,
            partial_rotary_factor=(cfg.get("partial_rotary_factor")
                                   or (cfg.get("rope_parameters") or {})
                                   .get("partial_rotary_factor")
                                   # MiniMax checkpoints name it rotary_dim
                                   or (cfg["rotary_dim"] / hd
                                       if cfg.get("rotary_dim") else 1.0)),
            attention_sinks=arch.startswith("GptOss"),
            sliding_window=(cfg.get("sliding_window") or 0)
            if (arch.startswith("GptOss") or arch.startswith("Gemma")
                or arch.startswith(("Olmo3", "Exaone4", "Cohere2")))
            else 0,
            layer_types=tuple(cfg["layer_types"])
            if cfg.get("layer_types") else (
                tuple("sliding_attention"
                      if (i + 1) % int(cfg.get("sliding_window_pattern", 6))
                      else "full_attention"
                      for i in range(cfg.get("num_hidden_layers", 0)))
                if arch.startswith("Gemma3") else None),
            moe_act="clamped_swiglu" if arch.startswith("GptOss") else "silu",
            moe_bias=arch.startswith("GptOss"),
            router_logit_bias=arch.startswith("GptOss"),
            o_proj_bias=bool(cfg.get("attention_bias"))
            if arch.startswith("GptOss") else False,
            q_lora_rank=(cfg.get("q_lora_rank") or 0)
            if arch.startswith("Deepseek") else 0,
            kv_lora_rank=(cfg.get("kv_lora_rank") or 0)
            if arch.startswith("Deepseek") else 0,
            qk_nope_head_dim=cfg.get("qk_nope_head_dim", 0) or 0,
            qk_rope_head_dim=cfg.get("qk_rope_head_dim", 0) or 0,
            v_head_dim=cfg.get("v_head_dim", 0) or 0,
            rope_interleave=bool(cfg.get("rope_interleave", True)),
            sandwich_norms=(arch.startswith("Gemma2")
                            or arch.startswith("Gemma3")
                            # GLM-4 dense: same four-norm layer flow,
                            # plain RMSNorm weights (no (1+w) offset)
                            or arch == "Glm4ForCausalLM"),
            norm_after=arch.startswith(("Olmo2", "Olmo3", "Exaone4")),
            qk_norm_full=arch.startswith(("Olmo2", "Olmo3", "Olmoe",
                                          "MiniMaxM2")),
            embed_scale=(cfg.get("hidden_size", 4096) ** 0.5
                         if arch.startswith("Gemma")
                         else (cfg.get("embedding_multiplier") or 0.0)
                         if arch.startswith("Granite") else 0.0),
            attn_logit_softcap=(cfg.get("attn_logit_softcapping") or 0.0)
            if arch.startswith("Gemma") else 0.0,
            final_logit_softcap=(cfg.get("final_logit_softcapping") or 0.0)
            if arch.startswith("Gemma") else 0.0,
            attn_scale=((cfg.get("query_pre_attn_scalar") or 0) ** -0.5
                        if (arch.startswith("Gemma")
                            and cfg.get("query_pre_attn_scalar"))
                        else (cfg.get("attention_multiplier") or 0.0)
                        if arch.startswith("Granite") else 0.0),
            residual_multiplier=(cfg.get("residual_multiplier") or 0.0)
            if arch.startswith("Granite") else 0.0,
            logits_scaling=(cfg.get("logits_scaling") or 0.0)
            if arch.startswith("Granite") else 0.0,
            rope_mode=("pairwise"
                       if (arch.startswith(("Ernie4_5", "Cohere",
                                            "Helium"))
                           or arch in ("GlmForCausalLM",
                                       "Glm4ForCausalLM"))
                       else "neox"),
            norm_type=("layernorm" if arch.startswith("Cohere")
                       else "rmsnorm"),
            parallel_block=arch.startswith("Cohere"),
            logits_multiplier=(cfg.get("logit_scale") or 0.0)
            if arch.startswith("Cohere") else 0.0,
            no_rope_layers=(tuple(cfg["no_rope_layers"])
                            if cfg.get("no_rope_layers") else (
                                # EXAONE-4 hybrid: global-NoPE — rope on
                                # sliding layers only, full layers unroped
                                tuple(1 if lt == "sliding_attention" else 0
                                      for lt in cfg["layer_types"])
                                if (arch.startswith(("Exaone4", "Cohere2"))
                                    and cfg.get("sliding_window")
                                    and cfg.get("layer_types")) else
                                tuple(0 if (i + 1) % 4 == 0 else 1
                                      for i in range(
                                          cfg.get("num_hidden_layers", 0)))
                                if arch.startswith("SmolLM3") else None)),
            rope_local_theta=(cfg.get("rope_local_base_freq") or 0.0)
            if arch.startswith("Gemma3")
            else (slide_theta if arch.startswith("Olmo3") else 0.0),
            mlp_act=("gelu_tanh" if arch.startswith("Gemma")
                     else "relu2" if arch.startswith("Arcee")
                     else "silu"),
            mlp_no_gate=arch.startswith("Arcee"),
        )

    @classmethod
    def from_dir(cls, model_dir: str | Path) -> "ModelSpec":
        with open(Path(model_dir) / "config.json") as f:
            return cls.from_hf_config(json.load(f))


# Named presets so the bench / tests / control plane can run with
# random-init weights and no network (BASELINE.json: synthetic data).
PRESETS: dict[str, ModelSpec] = {
    # DeepSeek-V3/R1 671B: MLA latent attention + 256-expert sigmoid+bias
    # grouped MoE (the R1 checkpoints share this architecture)
    "deepseek-v3": ModelSpec(
        architecture="DeepseekV3ForCausalLM", vocab_size=129280,
        hidden_size=7168, intermediate_size=18432, num_layers=61,
        num_heads=128, num_kv_heads=128, head_dim=192,
        rope_theta=10000.0, max_position_embeddings=163840,
        rope_scaling={"rope_type": "yarn", "factor": 40.0,
                      "beta_fast": 32.0, "beta_slow": 1.0,
                      "mscale": 1.0, "mscale_all_dim": 1.0,
                      "original_max_position_embeddings": 4096},
        eos_token_id=1, num_experts=256, num_experts_per_tok=8,
        moe_intermediate_size=2048, router_mode="sigmoid_bias",
        n_shared_experts=1, first_k_dense_replace=3,
        routed_scaling_factor=2.5, n_group=8, topk_group=4,
        norm_topk_prob=True,
        q_lora_rank=1536, kv_lora_rank=512, qk_nope_head_dim=128,
        qk_rope_head_dim=64, v_head_dim=128,
    ),

    "llama-3-8b": ModelSpec(),
    "llama-3-70b": ModelSpec(
        hidden_size=8192, intermediate_size=28672, num_layers=80,
        num_heads=64, num_kv_heads=8,
    ),
    # Llama-3.1: same graph as 3.0 plus llama3 rope scaling and 128k context
    "llama-3.1-8b": ModelSpec(
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072,
    ),
    "llama-3.1-70b": ModelSpec(
        hidden_size=8192, intermediate_size=28672, num_layers=80,
        num_heads=64, num_kv_heads=8,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072,
    ),
    "qwen3-32b": ModelSpec(
        architecture="Qwen3ForCausalLM", vocab_size=151936, hidden_size=5120,
        intermediate_size=25600, num_layers=64, num_heads=64, num_kv_heads=8,
        rope_theta=1000000.0, max_position_embeddings=40960, qk_norm=True,
        eos_token_id=151645,
    ),
    "qwen3-14b": ModelSpec(
        architecture="Qwen3ForCausalLM", vocab_size=151936, hidden_size=5120,
        intermediate_size=17408, num_layers=40, num_heads=40, num_kv_heads=8,
        rope_theta=1000000.0, max_position_embeddings=40960, qk_norm=True,
        eos_token_id=151645,
    ),
    "qwen2.5-7b": ModelSpec(
        architecture="Qwen2ForCausalLM", vocab_size=152064, hidden_size=3584,
        intermediate_size=18944, num_layers=28, num_heads=28, num_kv_heads=4,
        rope_theta=1000000.0, attention_bias=True, eos_token_id=151645,
    ),
    # Qwen3-30B-A3B: 128-expert top-8 MoE, 3B active params
    "qwen3-30b-a3b": ModelSpec(
        architecture="Qwen3MoeForCausalLM", vocab_size=151936,
        hidden_size=2048, intermediate_size=6144, num_layers=48,
        num_heads=32, num_kv_heads=4, head_dim=128, rope_theta=1000000.0,
        max_position_embeddings=40960, qk_norm=True, eos_token_id=151645,
        num_experts=128, num_experts_per_tok=8, moe_intermediate_size=768,
    ),
    # Qwen3-235B-A22B (the BASELINE.md 8-GPU MoE row): same Qwen3-MoE
    # graph as 30B-A3B, bigger dims — serves TP8 over xGMI
    "qwen3-235b-a22b": ModelSpec(
        architecture="Qwen3MoeForCausalLM", vocab_size=151936,
        hidden_size=4096, intermediate_size=12288, num_layers=94,
        num_heads=64, num_kv_heads=4, head_dim=128, rope_theta=1000000.0,
        max_position_embeddings=40960, qk_norm=True, eos_token_id=151645,
        num_experts=128, num_experts_per_tok=8, moe_intermediate_size=1536,
    ),
    # GLM-4.5-Air: Glm4Moe graph — sigmoid+bias
    # router, 1 shared expert, first layer dense, partial rotary 0.5
    "glm-4.5-air": ModelSpec(
        architecture="Glm4MoeForCausalLM", vocab_size=151552,
        hidden_size=4096, intermediate_size=10944, num_layers=46,
        num_heads=96, num_kv_heads=8, head_dim=128, rope_theta=1000000.0,
        max_position_embeddings=131072, qk_norm=True, eos_token_id=151329,
        num_experts=128, num_experts_per_tok=8, moe_intermediate_size=1408,
        router_mode="sigmoid_bias", n_shared_experts=1,
        first_k_dense_replace=1, partial_rotary_factor=0.5,
    ),
    # GPT-OSS 20B / 120B: 128/32-expert top-4 MoE,
    # alternating sliding(128)/full attention with sinks, clamped swiglu,
    # YaRN rope, head_dim 64. CPU-exact; CDNA4 kernels are an r3 item
    # (the GPU ops fail loudly on D=64/sinks/window).
    "gpt-oss-20b": ModelSpec(
        architecture="GptOssForCausalLM", vocab_size=201088,
        hidden_size=2880, intermediate_size=2880, num_layers=24,
        num_heads=64, num_kv_heads=8, head_dim=64, rope_theta=150000.0,
        rope_scaling={"rope_type": "yarn", "factor": 32.0,
                      "beta_fast": 32.0, "beta_slow": 1.0, "truncate": False,
                      "original_max_position_embeddings": 4096},
        max_position_embeddings=131072, eos_token_id=200002,
        attention_bias=True, o_proj_bias=True, attention_sinks=True,
        sliding_window=128, num_experts=32, num_experts_per_tok=4,
        moe_intermediate_size=2880, moe_act="clamped_swiglu", moe_bias=True,
        router_logit_bias=True, norm_topk_prob=True,
    ),
    "gpt-oss-120b": ModelSpec(
        architecture="GptOssForCausalLM", vocab_size=201088,
        hidden_size=2880, intermediate_size=2880, num_layers=36,
        num_heads=64, num_kv_heads=8, head_dim=64, rope_theta=150000.0,
        rope_scaling={"rope_type": "yarn", "factor": 32.0,
                      "beta_fast": 32.0, "beta_slow": 1.0, "truncate": False,
                      "original_max_position_embeddings": 4096},
        max_position_embeddings=131072, eos_token_id=200002,
        attention_bias=True, o_proj_bias=True, attention_sinks=True,
        sliding_window=128, num_experts=128, num_experts_per_tok=4,
        moe_intermediate_size=2880, moe_act="clamped_swiglu", moe_bias=True,
        router_logit_bias=True, norm_topk_prob=True,
    ),
    # Mixtral 8x7B: 8-expert top-2 MoE on the llama graph
    "mixtral-8x7b": ModelSpec(
        architecture="MixtralForCausalLM", vocab_size=32000,
        hidden_size=4096, intermediate_size=14336, num_layers=32,
        num_heads=32, num_kv_heads=8, rope_theta=1000000.0,
        max_position_embeddings=32768, eos_token_id=2,
        num_experts=8, num_experts_per_tok=2, moe_intermediate_size=14336,
    ),
    # tiny MoE (CPU-testable routing/dispatch plumbing)
    "tiny-moe": ModelSpec(
        vocab_size=512, hidden_size=128, intermediate_size=256, num_layers=2,
        num_heads=4, num_kv_heads=2, head_dim=32, max_position_embeddings=512,
        rope_theta=10000.0, eos_token_id=1,
        num_experts=8, num_experts_per_tok=2, moe_intermediate_size=64,
    ),
    # Mistral v0.3: llama-compatible compute graph (no SWA since v0.1;
    # GQA 32/8, theta 1e6) — runs on the same CDNA4 kernel set
    "mistral-7b": ModelSpec(
        architecture="MistralForCausalLM", vocab_size=32768, hidden_size=4096,
        intermediate_size=14336, num_layers=32, num_heads=32, num_kv_heads=8,
        rope_theta=1000000.0, max_position_embeddings=32768,
    ),
    # tiny CPU-testable model (OPT-125m-scale plumbing per BASELINE.json cfg 1)
    "tiny": ModelSpec(
        vocab_size=512, hidden_size=128, intermediate_size=256, num_layers=2,
        num_heads=4, num_kv_heads=2, head_dim=32, max_position_embeddings=512,
        rope_theta=10000.0, eos_token_id=1,
    ),
    # OLMo-2 13B (norm-after layers + full-projection qk-norm)
    "olmo-2-13b": ModelSpec(
        architecture="Olmo2ForCausalLM", vocab_size=100352,
        hidden_size=5120, intermediate_size=13824, num_layers=40,
        num_heads=40, num_kv_heads=40, head_dim=128,
        rope_theta=500000.0, max_position_embeddings=4096,
        rms_norm_eps=1e-6, eos_token_id=100257, qk_norm=True,
        norm_after=True, qk_norm_full=True,
    ),
    # Phi-4 14B (Phi3ForCausalLM graph: fused qkv/gate_up, plain rope)
    "phi-4": ModelSpec(
        architecture="Phi3ForCausalLM", vocab_size=100352,
        hidden_size=5120, intermediate_size=17920, num_layers=40,
        num_heads=40, num_kv_heads=10, head_dim=128, rope_theta=250000.0,
        max_position_embeddings=16384, eos_token_id=100257,
    ),
    # DeepSeek-R1: identical architecture/dims to V3 (reasoning ckpt)
    "deepseek-r1": ModelSpec(
        architecture="DeepseekV3ForCausalLM", vocab_size=129280,
        hidden_size=7168, intermediate_size=18432, num_layers=61,
        num_heads=128, num_kv_heads=128, head_dim=192,
        rope_theta=10000.0, max_position_embeddings=163840,
        rope_scaling={"rope_type": "yarn", "factor": 40.0,
                      "beta_fast": 32.0, "beta_slow": 1.0,
                      "mscale": 1.0, "mscale_all_dim": 1.0,
                      "original_max_position_embeddings": 4096},
        eos_token_id=1, num_experts=256, num_experts_per_tok=8,
        moe_intermediate_size=2048, router_mode="sigmoid_bias",
        n_shared_experts=1, first_k_dense_replace=3,
        routed_scaling_factor=2.5, n_group=8, topk_group=4,
        norm_topk_prob=True,
        q_lora_rank=1536, kv_lora_rank=512, qk_nope_head_dim=128,
        qk_rope_head_dim=64, v_head_dim=128,
    ),
    # Kimi-K2 1T: DeepseekV3 architecture at larger expert count
    "kimi-k2": ModelSpec(
        architecture="DeepseekV3ForCausalLM", vocab_size=163840,
        hidden_size=7168, intermediate_size=18432, num_layers=61,
        num_heads=64, num_kv_heads=64, head_dim=192,
        rope_theta=50000.0, max_position_embeddings=131072,
        eos_token_id=163585, num_experts=384, num_experts_per_tok=8,
        moe_intermediate_size=2048, router_mode="sigmoid_bias",
        n_shared_experts=1, first_k_dense_replace=1,
        routed_scaling_factor=2.827, n_group=1, topk_group=1,
        norm_topk_prob=True,
        q_lora_rank=1536, kv_lora_rank=512, qk_nope_head_dim=128,
        qk_rope_head_dim=64, v_head_dim=128,
    ),
    # Gemma-2: sandwich norms, GeGLU, softcapping, alternating SWA,
    # scaled tied embeddings (CPU-oracle family; GPU kernels share the
    # GPT-OSS r3 window work — head_dim 256 needs a D-template too)
    "gemma-2-9b": ModelSpec(
        architecture="Gemma2ForCausalLM", vocab_size=256000,
        hidden_size=3584, intermediate_size=14336, num_layers=42,
        num_heads=16, num_kv_heads=8, head_dim=256, rope_theta=10000.0,
        max_position_embeddings=8192, tie_word_embeddings=True,
        rms_norm_eps=1e-6, eos_token_id=1, sliding_window=4096,
        sandwich_norms=True, embed_scale=3584 ** 0.5,
        attn_logit_softcap=50.0, final_logit_softcap=30.0,
        attn_scale=224 ** -0.5, mlp_act="gelu_tanh",
    ),
    # Gemma-3: dual rope (local 10k on sliding layers, linear-scaled 1M
    # global), (1+w) qk-norm, 5:1 SWA pattern, no softcapping
    "gemma-3-27b": ModelSpec(
        architecture="Gemma3ForCausalLM", vocab_size=262208,
        hidden_size=5376, intermediate_size=21504, num_layers=62,
        num_heads=32, num_kv_heads=16, head_dim=128,
        rope_theta=1000000.0, rope_local_theta=10000.0,
        rope_scaling={"rope_type": "linear", "factor": 8.0},
        max_position_embeddings=131072, tie_word_embeddings=True,
        rms_norm_eps=1e-6, eos_token_id=1, sliding_window=1024,
        layer_types=tuple("sliding_attention" if (i + 1) % 6 else
                          "full_attention" for i in range(62)),
        sandwich_norms=True, qk_norm=True, embed_scale=5376 ** 0.5,
        attn_scale=168 ** -0.5, mlp_act="gelu_tanh",
    ),
    # CPU-test preset: Gemma-shaped sandwich-norm layers (tests the
    # residual-None PP handoff + softcap/GeGLU paths on gloo)
    "tiny-gemma": ModelSpec(
        architecture="Gemma2ForCausalLM", vocab_size=512, hidden_size=128,
        intermediate_size=256, num_layers=4, num_heads=4, num_kv_heads=2,
        head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
        rms_norm_eps=1e-6, eos_token_id=1, tie_word_embeddings=True,
        sliding_window=8, sandwich_norms=True, embed_scale=128 ** 0.5,
        attn_logit_softcap=50.0, final_logit_softcap=30.0,
        attn_scale=24 ** -0.5, mlp_act="gelu_tanh",
    ),
    # CPU-test preset: DeepSeek-shaped MLA + MoE (tests/test_deepseek.py,
    # TP exactness on gloo)
    "tiny-mla": ModelSpec(
        architecture="DeepseekV3ForCausalLM", vocab_size=512,
        hidden_size=128, intermediate_size=256, num_layers=2,
        num_heads=4, num_kv_heads=4, head_dim=48,
        max_position_embeddings=256, rope_theta=10000.0, eos_token_id=1,
        num_experts=8, num_experts_per_tok=2, moe_intermediate_size=64,
        router_mode="sigmoid_bias", n_shared_experts=1,
        first_k_dense_replace=1, routed_scaling_factor=1.5, n_group=2,
        topk_group=1, q_lora_rank=64, kv_lora_rank=96,
        qk_nope_head_dim=32, qk_rope_head_dim=16, v_head_dim=32,
    ),
}


@dataclass
class EngineConfig:
    model: str = "llama-3-8b"              # preset name or model dir
    dtype: str = "bfloat16"
    kv_cache_dtype: str = "bf16"           # bf16 | fp8 (e4m3, halves KV bytes)
    block_size: int = 16
    gpu_memory_utilization: float = 0.90
    max_num_seqs: int = 256
    max_prefill_tokens: int = 8192         # per-step prefill token budget
    max_model_len: int = 8192
    kv_cache_blocks: int | None = None     # override (CPU tests)
    tp_size: int = 1
    tp_rank: int = 0
    device: str = "cuda"
    seed: int = 0
    enforce_random_weights: bool = True    # no checkpoint: random init
    model_dir: str | None = None
    # host-DRAM KV offload tier (extended_kv_cache in the reference schema)
    kv_offload_gb: float = 0.0
    # automatic prefix caching: cache-hit prompts recompute only the suffix
    # through the paged prefill-with-history MFMA kernel; the r1 cap (the
    # old paged-decode-row fallback cost ~3.5x per token) is retired — any
    # suffix length now runs at flash-prefill efficiency
    enable_prefix_caching: bool = False
    prefix_cache_suffix_cap: int = 1 << 30
    # chunked prefill (reference: vLLM --enable-chunked-prefill): a prompt
    # longer than max_prefill_tokens is split into budget-sized chunks
    # (chunk 0 = capped prefill batch, continuations = paged prefill-with-
    # history tiles), alternating 1:1 with decode steps so running
    # sequences keep a bounded time-between-tokens during long-prompt
    # admission. Default ON since r2: continuations run at flash-prefill
    # efficiency (the r1 3.5x continuation penalty is gone)
    enable_chunked_prefill: bool = True
    # speculative decoding (reference speculative_config schema):
    # {"method": "ngram", "num_draft_tokens": 3, "ngram_max": 3, "ngram_min": 1}
    speculative: dict | None = None
    # W4 runtime quantization ("w4"): the big serving weights are packed
    # int4 in HBM after load and decode GEMMs dequantize in-register
    # (ops/csrc/w4_gemm.hip) — 70B-class models stay resident packed
    # instead of inflating to bf16 (reference: vLLM --quantization)
    quantize_runtime: str | None = None
    # CPU weight offload (reference: vLLM --cpu-offload-gb, the GGUF
    # partial-offload placement): ~this many GiB of trailing layers'
    # weights live pinned in host DRAM and stream to a double-buffered
    # device staging area one layer ahead of use (engine/offload.py)
    cpu_offload_gb: float = 0.0
    # uneven pipeline-stage layer counts (len == pp_size, sums to
    # num_layers) — the native analog of the reference's per-GPU GGUF
    # tensor_split proportions; None = even split (remainder to early
    # stages so the sampling stage carries less)
    pp_partition: list[int] | None = None
    # LoRA adapters merged into the weights at load (reference lora_list)
    lora_dirs: list[str] = field(default_factory=list)
    # GGUF checkpoint execution: dequantized to bf16 at load (utils/gguf.py)
    gguf_path: str | None = None
    # admission hysteresis: open a prefill step only when this many requests
    # wait, one has waited admission_max_wait_s, or nothing is running —
    # keeps steady-state decode on the hipGraph path instead of degrading
    # every step to a tiny eager mixed batch
    admission_min_seqs: int = 16
    admission_max_wait_s: float = 0.1

    spec: ModelSpec = field(default_factory=ModelSpec)

    def __post_init__(self):
        if self.model in PRESETS:
            import dataclasses

            # copy: configs may tweak their spec (tests shrink num_layers,
            # qk_norm overrides) and must not mutate the shared preset
            self.spec = dataclasses.replace(PRESETS[self.model])
        elif self.model_dir:
            self.spec = ModelSpec.from_dir(self.model_dir)
        elif self.model.endswith(".gguf") and Path(self.model).is_file():
            from ..utils.gguf import spec_from_gguf

            self.gguf_path = self.model
            self.spec = spec_from_gguf(self.model)
        elif Path(self.model).is_dir():
            self.model_dir = self.model
            self.spec = ModelSpec.from_dir(self.model)
        self.max_model_len = min(self.max_model_len, self.spec.max_position_embeddings)
        if self.enable_chunked_prefill and self.speculative and \
                self.speculative.get("method") in ("eagle", "eagle3", "mtp"):
            # draft-model speculation seeds from full-prompt prefill hiddens,
            # which chunked admission does not produce in one step; chunked
            # prefill (default-on since r2) yields to the explicit
            # speculative config
            import logging

            logging.getLogger(__name__).info(
                "disabling chunked prefill: incompatible with draft-model "
                "speculative decoding")
            self.enable_chunked_prefill = False
