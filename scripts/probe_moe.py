import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import random
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

cfg = EngineConfig(model="qwen3-30b-a3b", device="cuda:0", max_model_len=4096,
                   max_num_seqs=128, seed=0)
t0 = time.perf_counter()
eng = LLMEngine(cfg)
print(f"init {time.perf_counter()-t0:.1f}s", flush=True)
rng = random.Random(0)
p = SamplingParams(max_tokens=64, ignore_eos=True)
for _ in range(128):
    eng.add_request([rng.randrange(2, 100000) for _ in range(256)], p)
for i in range(40):
    t = time.perf_counter()
    outs = eng.step()
    print(f"step {i}: {1000*(time.perf_counter()-t):.1f} ms, outs={len(outs)}", flush=True)
