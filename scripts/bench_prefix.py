"""A/B automatic prefix caching on a shared-prefix serving workload
(chat-style: 1, 900-token shared system prompt + ~100 unique tokens)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams  # noqa: E402


def run(caching: bool, n_req=64, prefix_len=1900, unique=100, osl=64):
    eng = LLMEngine(EngineConfig(
        model="llama-3-8b", device="cuda", max_model_len=4096,
        max_num_seqs=64, enable_prefix_caching=caching,
        prefix_cache_suffix_cap=256,
    ))
    g = torch.Generator().manual_seed(0)
    prefix = torch.randint(10, 100000, (prefix_len,), generator=g).tolist()
    p = SamplingParams(max_tokens=osl, ignore_eos=True)
    # warm the cache with one request
    eng.generate([prefix + torch.randint(10, 100000, (unique,),
                                         generator=g).tolist()], p)
    reqs = [prefix + torch.randint(10, 100000, (unique,),
                                   generator=g).tolist()
            for _ in range(n_req)]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for q in reqs:
        eng.add_request(q, p)
    ttfts, done = {}, 0
    while eng.has_unfinished():
        for out in eng.step():
            if out.request_id not in ttfts:
                ttfts[out.request_id] = time.perf_counter() - t0
            if out.finished:
                done += 1
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    tps = n_req * osl / el
    hits = getattr(eng.scheduler.kv.allocator, "hits", 0)
    lat = sorted(ttfts.values())
    print(f"caching={caching}: {tps:8.0f} out-tok/s  elapsed {el:5.2f}s  "
          f"p50_ttft {lat[len(lat)//2]*1000:7.0f}ms  cache_hits {hits}")
    del eng
    torch.cuda.empty_cache()
    return tps


if __name__ == "__main__":
    a = run(False)
    b = run(True)
    print(f"speedup {b/a:.2f}x")
