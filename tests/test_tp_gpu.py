"""Multi-GPU TP smoke (rehearsed so zero lease-minutes are
wasted the day a multi-GPU box appears). Skips on <2 visible devices —
single-GPU gpurun leases run everything else; an 8-GPU node runs this
automatically via the normal `pytest -m gpu` invocation."""
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not torch.cuda.is_available()
                    or torch.cuda.device_count() < 2,
                    reason="needs >=2 GPUs")
def test_tp2_engine_matches_tp1():
    """Spawn a 2-rank RCCL TP group and check output equality vs TP1.

    Runs as subprocesses (one per GPU) exactly like the serve manager's
    engine launch; gloo-based equivalents run on CPU in test_tp_cpu.py."""
    script = r"""
import os, sys, json
import torch
rank = int(os.environ["RANK"])
from gpustack_amd.parallel import init_tp
comm = init_tp(2, rank, master_port=29611, device_id=rank)
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
cfg = EngineConfig(model="llama-3-8b", device=f"cuda:{rank}",
                   max_model_len=512, max_num_seqs=8,
                   gpu_memory_utilization=0.2, tp_size=2, tp_rank=rank)
cfg.spec.num_layers = 4
eng = LLMEngine(cfg, comm)
prompts = [[1, 2, 3, 4, 5] * 8, [7, 6, 5] * 5]
p = SamplingParams(max_tokens=8, ignore_eos=True)
if rank == 0:
    out = eng.generate(prompts, p)
    print("TP2OUT:" + json.dumps(out), flush=True)
else:
    while eng.tp_active():
        eng.step()
"""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    procs = []
    for r in range(2):
        e = dict(env, RANK=str(r), WORLD_SIZE="2")
        procs.append(subprocess.Popen([sys.executable, "-c", script], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    outs = [p.communicate(timeout=600)[0] for p in procs]
    assert all(p.returncode == 0 for p in procs), outs
    import json as _json

    line = next(ln for ln in outs[0].splitlines() if ln.startswith("TP2OUT:"))
    tp2 = _json.loads(line[len("TP2OUT:"):])

    # TP1 reference in-process (random-init weights are TP-consistent by
    # construction: full tensors generated then sliced)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    cfg = EngineConfig(model="llama-3-8b", device="cuda:0", max_model_len=512,
                       max_num_seqs=8, gpu_memory_utilization=0.2)
    cfg.spec.num_layers = 4
    eng = LLMEngine(cfg)
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    tp1 = eng.generate([[1, 2, 3, 4, 5] * 8, [7, 6, 5] * 5], p)
    assert tp2 == tp1
