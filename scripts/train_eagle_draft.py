#!/usr/bin/env python3
"""Distill an EAGLE draft head from a (random-init) target model ON DEVICE.

The air-gapped environment has no published EAGLE checkpoints, and an
untrained draft gives acceptance ~0 (profiles/r03). This script produces a
REAL draft the honest way — the way EAGLE heads are produced in practice:
teacher-forced distillation against the target's own greedy trajectories.

  1. roll out greedy continuations from the serving engine (the exact
     decode-time distribution the draft must imitate),
  2. collect the target's post-final-norm hiddens over those sequences,
  3. train the draft (fc + one decoder layer + norm, sharing the target's
     embedding/lm_head/rope) with cross-entropy against the target's next
     token, using a differentiable mirror of engine/eagle.py's forward,
  4. save an EAGLE-release-named safetensors checkpoint that the engine's
     existing `speculative_config.draft_dir` loader consumes.

    python scripts/train_eagle_draft.py --out /tmp/draft [--steps 400]
    python scripts/train_eagle_draft.py --selftest   # CPU equivalence check
"""
from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import gpustack_amd.engine  # noqa: F401  (import order)
import torch
import torch.nn.functional as F


def rope_neox(x, cos_sin, positions):
    """Differentiable mirror of ops.rotary_embedding (neox style)."""
    T, H, D = x.shape
    half = D // 2
    cs = cos_sin[positions]            # [T, D] f32
    cos = cs[:, :half].unsqueeze(1)
    sin = cs[:, half:].unsqueeze(1)
    x1, x2 = x[..., :half], x[..., half:]
    return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)


class DraftTrainer(torch.nn.Module):
    """fp32 trainable mirror of EagleProposer's draft head."""

    def __init__(self, spec, device):
        super().__init__()
        h, d = spec.hidden_size, spec.head_dim
        self.spec = spec
        self.hq, self.hkv = spec.num_heads, spec.num_kv_heads
        self.eps = spec.rms_norm_eps
        def P(*shape):
            return torch.nn.Parameter(torch.randn(*shape, device=device) * 0.02)
        self.fc = P(h, 2 * h)
        self.q = P(self.hq * d, h)
        self.k = P(self.hkv * d, h)
        self.v = P(self.hkv * d, h)
        self.o = P(h, self.hq * d)
        self.gate = P(spec.intermediate_size, h)
        self.up = P(spec.intermediate_size, h)
        self.down = P(h, spec.intermediate_size)
        self.input_norm = torch.nn.Parameter(torch.ones(h, device=device))
        self.post_norm = torch.nn.Parameter(torch.ones(h, device=device))
        self.final_norm = torch.nn.Parameter(torch.ones(h, device=device))

    def rms(self, x, w):
        return x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps) * w

    def forward(self, emb, h_prev, cos_sin, positions):
        """Rows [T, h]: draft hidden for each position (causal self-attn).

        Mirrors engine/eagle.py: x = fc([emb; h_prev]); DecoderLayer with
        residual=None semantics; final fused_add_rms_norm."""
        d = self.spec.head_dim
        x = F.linear(torch.cat([emb, h_prev], dim=-1), self.fc)
        residual = x
        h = self.rms(x, self.input_norm)
        T = h.shape[0]
        q = F.linear(h, self.q).view(T, self.hq, d)
        k = F.linear(h, self.k).view(T, self.hkv, d)
        v = F.linear(h, self.v).view(T, self.hkv, d)
        q = rope_neox(q, cos_sin, positions)
        k = rope_neox(k, cos_sin, positions)
        gq = self.hq // self.hkv
        k = k.repeat_interleave(gq, dim=1)
        v = v.repeat_interleave(gq, dim=1)
        att = F.scaled_dot_product_attention(
            q.permute(1, 0, 2), k.permute(1, 0, 2), v.permute(1, 0, 2),
            is_causal=True)
        a = F.linear(att.permute(1, 0, 2).reshape(T, -1), self.o)
        residual = residual + a
        h2 = self.rms(residual, self.post_norm)
        m = F.linear(F.silu(F.linear(h2, self.gate)) * F.linear(h2, self.up),
                     self.down)
        return self.rms(residual + m, self.final_norm)

    def save(self, out_dir: str) -> None:
        from safetensors.torch import save_file

        out = Path(out_dir)
        out.mkdir(parents=True, exist_ok=True)
        t = {
            "fc.weight": self.fc.data,
            "layers.0.self_attn.q_proj.weight": self.q.data,
            "layers.0.self_attn.k_proj.weight": self.k.data,
            "layers.0.self_attn.v_proj.weight": self.v.data,
            "layers.0.self_attn.o_proj.weight": self.o.data,
            "layers.0.mlp.gate_proj.weight": self.gate.data,
            "layers.0.mlp.up_proj.weight": self.up.data,
            "layers.0.mlp.down_proj.weight": self.down.data,
            "layers.0.input_layernorm.weight": self.input_norm.data,
            "layers.0.post_attention_layernorm.weight": self.post_norm.data,
            "norm.weight": self.final_norm.data,
        }
        save_file({k: v.to(torch.bfloat16).cpu().contiguous()
                   for k, v in t.items()}, str(out / "draft.safetensors"))


def collect_hiddens(runner, tokens: list[int], device):
    """Target post-final-norm hiddens for one sequence (KV writes skipped)."""
    from gpustack_amd import ops
    from gpustack_amd.models.llama import ForwardMeta

    L = len(tokens)
    tok = torch.tensor(tokens, dtype=torch.long, device=device)
    pos = torch.arange(L, dtype=torch.long, device=device)
    slots = torch.full((L,), -1, dtype=torch.long, device=device)
    tiles = ops.build_prefill_tiles([L], device)
    meta = ForwardMeta(is_prefill=True, positions=pos, slot_mapping=slots,
                       logits_indices=torch.zeros(1, dtype=torch.long,
                                                  device=device),
                       seq_lens_list=[L], tile_start=tiles[0],
                       tile_q0=tiles[1], tile_len=tiles[2])
    with torch.inference_mode():
        h = runner.model(tok, meta, runner.kv, return_hidden=True)
    return h.float().clone()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="/tmp/eagle_draft")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--device", default=None)
    ap.add_argument("--rollouts", type=int, default=256)
    ap.add_argument("--isl", type=int, default=256, help="max prompt len (prompts vary 32..isl)")
    ap.add_argument("--osl", type=int, default=160)
    ap.add_argument("--steps", type=int, default=400)
    ap.add_argument("--val", type=int, default=16, help="held-out sequences")
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--selftest", action="store_true")
    args = ap.parse_args()

    if args.selftest:
        return selftest()

    import random

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    dev = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    cfg = EngineConfig(model=args.model, device=dev, max_model_len=1024,
                       max_num_seqs=max(8, args.rollouts), seed=0,
                       gpu_memory_utilization=0.45)
    eng = LLMEngine(cfg)
    rng = random.Random(77)
    vocab = cfg.spec.vocab_size
    t0 = time.time()
    prompts = [[rng.randrange(2, vocab)
                for _ in range(rng.randrange(32, args.isl + 1))]
               for _ in range(args.rollouts)]
    outs = eng.generate(prompts, SamplingParams(max_tokens=args.osl,
                                                ignore_eos=True))
    seqs = [p + o for p, o in zip(prompts, outs)]
    print(f"rollouts done in {time.time()-t0:.1f}s", flush=True)

    runner = eng.runner
    model = runner.model
    spec = cfg.spec
    data = []
    for sq in seqs:
        H = collect_hiddens(runner, sq, dev)
        data.append((torch.tensor(sq, dtype=torch.long, device=dev), H))
    val = data[:args.val]
    data = data[args.val:]
    print(f"hiddens collected ({len(data)} train / {len(val)} val)", flush=True)

    draft = DraftTrainer(spec, dev)
    opt = torch.optim.Adam(draft.parameters(), lr=args.lr)
    embed = model.embed.float()
    lm_head = model.lm_head.float()
    cos_sin = model.cos_sin.float()
    t0 = time.time()
    for step in range(args.steps):
        toks, H = data[step % len(data)]
        L = toks.shape[0]
        # draft row p (p>=1): input (emb(t_p), H_{p-1}) -> predict t_{p+1}
        emb = F.embedding(toks[1:L - 1], embed)
        h_prev = H[0:L - 2]
        pos = torch.arange(1, L - 1, device=dev)
        out = draft(emb, h_prev, cos_sin, pos)
        logits = F.linear(out, lm_head)
        loss = F.cross_entropy(logits, toks[2:L])
        opt.zero_grad()
        loss.backward()
        opt.step()
        if step % 100 == 0 or step == args.steps - 1:
            with torch.no_grad():
                acc = (logits.argmax(-1) == toks[2:L]).float().mean().item()
                vac, vn = 0.0, 0
                for vt, vH in val:
                    Lv = vt.shape[0]
                    vo = draft(F.embedding(vt[1:Lv - 1], embed), vH[0:Lv - 2],
                               cos_sin, torch.arange(1, Lv - 1, device=dev))
                    vl = F.linear(vo, lm_head)
                    vac += (vl.argmax(-1) == vt[2:Lv]).float().sum().item()
                    vn += Lv - 2
            print(f"step {step}: loss {loss.item():.3f} train-acc {acc:.3f} "
                  f"VAL-acc {vac/max(vn,1):.3f} ({time.time()-t0:.0f}s)",
                  flush=True)
    draft.save(args.out)
    print(f"saved draft checkpoint to {args.out}", flush=True)


def selftest():
    """CPU equivalence: DraftTrainer forward == EagleProposer._forward_prefill
    on shared random weights (tiny model)."""
    from gpustack_amd.engine import EngineConfig, LLMEngine

    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       speculative={"method": "eagle", "num_draft_tokens": 2})
    eng = LLMEngine(cfg)
    eagle = eng.runner.eagle
    spec = cfg.spec
    dt = DraftTrainer(spec, "cpu")
    # copy the proposer's random-init weights into the trainer
    la = eagle.layers[0].attn
    nq = spec.num_heads * spec.head_dim
    nk = spec.num_kv_heads * spec.head_dim
    dt.fc.data = eagle.fc_ws[0].float()
    dt.q.data = la.qkv_w[:nq].float()
    dt.k.data = la.qkv_w[nq:nq + nk].float()
    dt.v.data = la.qkv_w[nq + nk:].float()
    dt.o.data = la.o_w.float()
    i = spec.intermediate_size
    dt.gate.data = eagle.layers[0].mlp.gate_up_w[:i].float()
    dt.up.data = eagle.layers[0].mlp.gate_up_w[i:].float()
    dt.down.data = eagle.layers[0].mlp.down_w.float()
    dt.input_norm.data = eagle.layers[0].input_norm.float()
    dt.post_norm.data = eagle.layers[0].post_attn_norm.float()
    dt.final_norm.data = eagle.norms[0].float()

    torch.manual_seed(0)
    L = 12
    toks = torch.randint(2, spec.vocab_size, (L,))
    h_prev = torch.randn(L, spec.hidden_size) * 0.1
    model = eng.runner.model
    # proposer path (prefill over the draft KV)
    from gpustack_amd.engine.eagle import _DraftState

    st = eagle.states.setdefault("t", _DraftState())
    eagle._ensure_blocks(st, L)
    slots = torch.tensor([eagle._slot(st, p) for p in range(L)],
                         dtype=torch.long)
    pos = torch.arange(L, dtype=torch.long)
    ref = eagle._forward_prefill(toks, h_prev.to(model.dtype), pos, slots, [L])
    # trainer path
    emb = F.embedding(toks, model.embed).float()
    got = dt(emb, h_prev, model.cos_sin.float(), pos)
    err = (got - ref.float()).abs().max().item()
    scale = ref.float().abs().max().item()
    print(f"selftest max abs err {err:.4f} (ref scale {scale:.3f})")
    assert err / max(scale, 1e-6) < 0.06, "trainer forward diverges from proposer"
    print("selftest OK")


if __name__ == "__main__":
    main()
