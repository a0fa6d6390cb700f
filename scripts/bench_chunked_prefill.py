"""Chunked-prefill A/B: inter-token stall of a running decode while a long
prompt is admitted (the metric chunked prefill exists to bound).

Runs one decode stream, then injects long prompts; reports the decode
stream's max inter-token gap and the long prompt's TTFT, with and without
chunked prefill. Evidence for profiles/ (run via gpurun on an MI355X)."""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def run(enable_chunked: bool, isl: int, budget: int, model: str) -> dict:
    import torch

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    cfg = EngineConfig(
        model=model,
        device="cuda" if torch.cuda.is_available() else "cpu",
        max_model_len=max(8192, isl + 256),
        max_num_seqs=64,
        max_prefill_tokens=budget,
        enable_chunked_prefill=enable_chunked,
        gpu_memory_utilization=0.85,
    )
    eng = LLMEngine(cfg)
    p_decode = SamplingParams(max_tokens=100000, ignore_eos=True)
    stream = eng.add_request([11, 12, 13, 14], p_decode)
    # warm the decode stream
    for _ in range(32):
        eng.step()
    gaps = []
    ttfts = {}
    last_emit = time.perf_counter()
    long_ids = []
    t_submit = {}
    for k in range(4):  # four long prompts arrive while decoding
        rid = eng.add_request(
            [(7 * t + k) % (cfg.spec.vocab_size - 16) for t in range(isl)],
                              SamplingParams(max_tokens=8, ignore_eos=True))
        long_ids.append(rid)
        t_submit[rid] = time.perf_counter()
        # drive until this prompt emits its first token
        while rid not in ttfts:
            outs = eng.step()
            now = time.perf_counter()
            for o in outs:
                if o.request_id == stream:
                    gaps.append(now - last_emit)
                    last_emit = now
                elif o.request_id in t_submit and o.request_id not in ttfts:
                    ttfts[o.request_id] = now - t_submit[o.request_id]
    eng.abort_request(stream)
    while eng.has_unfinished():
        eng.step()
    gaps.sort()
    return {
        "chunked": enable_chunked,
        "decode_max_gap_ms": round(max(gaps) * 1000, 2),
        "decode_p99_gap_ms": round(gaps[int(len(gaps) * 0.99) - 1] * 1000, 2),
        "decode_p50_gap_ms": round(gaps[len(gaps) // 2] * 1000, 2),
        "long_prompt_ttft_ms": [round(ttfts[r] * 1000, 2) for r in long_ids],
        "isl": isl, "budget": budget,
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--isl", type=int, default=6144)
    ap.add_argument("--budget", type=int, default=1024)
    ap.add_argument("--model", default="llama-3-8b")
    args = ap.parse_args()
    for chunked in (False, True):
        print(json.dumps(run(chunked, args.isl, args.budget, args.model)),
              flush=True)


if __name__ == "__main__":
    main()
