import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch
from gpustack_amd import ops

torch.manual_seed(0)
dev = "cuda"
E, K, H, I = 128, 8, 2048, 768
T = 4096
x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) / 8
gate_up_w = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=dev) / 16
down_w = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) / 16
logits = torch.randn(T, E, device=dev)
weights, experts = torch.topk(torch.softmax(logits, -1), K, dim=-1)
flat_exp = experts.reshape(-1)
flat_tok = torch.arange(T, device=dev).repeat_interleave(K)
flat_w = weights.reshape(-1).to(torch.bfloat16)
TK = flat_exp.numel()

def ck(name):
    torch.cuda.synchronize()
    print("ok:", name, flush=True)

order = torch.argsort(flat_exp, stable=True)
s_exp = flat_exp[order]
s_tok = flat_tok[order]
counts = torch.bincount(s_exp, minlength=E)
cap = int(counts.max())
ck(f"routing cap={cap}")
offs = counts.cumsum(0) - counts
pos = torch.arange(TK, device=dev) - offs[s_exp]
ck("pos")
xpad = x.new_zeros(E, cap, H)
xpad[s_exp, pos] = x[s_tok]
ck("xpad scatter")
gu = torch.bmm(xpad, gate_up_w.transpose(1, 2))
ck("bmm1")
act = torch.empty(E * cap, I, dtype=x.dtype, device=dev)
ops.silu_and_mul(act, gu.reshape(E * cap, 2 * I))
ck("silu")
hd = torch.bmm(act.view(E, cap, I), down_w.transpose(1, 2))
ck("bmm2")
contrib = x.new_zeros(TK, H)
contrib[order] = hd[s_exp, pos] * flat_w[order].unsqueeze(1)
ck("combine")
print("ALL OK")
