#!/usr/bin/env python3
"""Standalone timing of the fused MoE kernels at the qwen3-30b-a3b decode
shape vs the padded-bmm fallback."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from gpustack_amd import ops

def t(fn, n=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

T, E, K, H, I = 128, 128, 8, 2048, 768
dev = "cuda"
torch.manual_seed(0)
x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) / 8
w_gu = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=dev) / 16
w_d = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) / 16
logits = torch.randn(T, E, device=dev)
weights, experts = torch.topk(torch.softmax(logits, -1), K, dim=-1)
flat_exp = experts.reshape(-1)
flat_tok = torch.arange(T, device=dev).repeat_interleave(K)
flat_w = weights.reshape(-1).float()
order = torch.argsort(flat_exp, stable=True)
s_tok = flat_tok[order].to(torch.int32)
counts = torch.zeros(E, dtype=torch.int32, device=dev)
counts.scatter_add_(0, flat_exp, torch.ones_like(flat_exp, dtype=torch.int32))
offs = (counts.cumsum(0, dtype=torch.int32) - counts).to(torch.int32)
hip = ops._load_hip()
TK = T * K
act = x.new_empty(TK, I)
contrib = x.new_empty(TK, H)
order32 = order.to(torch.int32)

print("gate_up_us", round(t(lambda: hip.moe_gate_up_silu(act, x, w_gu, s_tok, offs, counts)), 1))
print("down_us", round(t(lambda: hip.moe_down_scale(contrib, act, w_d, offs, counts, order32, flat_w)), 1))
print("routing_us", round(t(lambda: (torch.argsort(flat_exp, stable=True),
                                     torch.zeros(E, dtype=torch.int32, device=dev).scatter_add_(0, flat_exp, torch.ones_like(flat_exp, dtype=torch.int32)))), 1))
# memory floor: full expert banks at 8 TB/s
gb = (w_gu.numel() + w_d.numel()) * 2 / 1e9
print("weight_gb_per_layer", round(gb, 3), "floor_us", round(gb / 8e12 * 1e15, 1))
