#!/usr/bin/env python3
"""Flagship serving benchmark (driver contract).

Measures the BASELINE.json metric — output tokens/sec (+ p50 TTFT) for
Llama-3-8B serving on MI355X — with one engine replica per GPU (the
reference's headline deployment shape: N replicas bin-packed on N GPUs,
BASELINE.json config 3). Synthetic random-token requests, random-init bf16
weights (no network), closed-loop at fixed concurrency so the continuous
batcher runs saturated; a "step" is one engine iteration (one varlen
prefill batch or one decode batch over all running sequences).

Usage:  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this via torch.distributed.run with one rank
per GPU; ranks are independent replicas (weak scaling) synchronized only
at the timing barriers.
"""
from __future__ import annotations

import argparse
import json
import os
import random
import time

import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=192)
    ap.add_argument("--warmup", type=int, default=48)
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--concurrency", type=int, default=512, help="target in-flight requests per GPU")
    ap.add_argument("--isl", type=int, default=256, help="synthetic prompt length")
    ap.add_argument("--osl", type=int, default=128, help="max output tokens per request")
    ap.add_argument("--max-model-len", type=int, default=4096)
    ap.add_argument("--speculative", default=None, choices=["ngram", "eagle", "eagle3"],
                    help="speculative decoding method (BASELINE config 5)")
    ap.add_argument("--draft-tokens", type=int, default=3)
    ap.add_argument("--draft-dir", default=None,
                    help="EAGLE/MTP draft checkpoint dir (speculative_config "
                         "draft_dir; produce one offline with "
                         "scripts/train_eagle_draft.py)")
    ap.add_argument("--prefix-caching", action="store_true",
                    help="enable automatic prefix caching (shared-prefix workloads)")
    ap.add_argument("--quantize", default=None, choices=["w4"],
                    help="runtime weight quantization (packed int4 in HBM)")
    ap.add_argument("--kv-dtype", default="bf16", choices=["bf16", "fp8"],
                    help="KV cache dtype (fp8 e4m3 halves KV bytes; compute stays bf16)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor-parallel degree: world becomes ONE replica "
                         "sharded over RCCL/xGMI (default: dp replicas)")
    ap.add_argument("--cp", type=int, default=1,
                    help="prefill-context-parallel degree (rows chunked "
                         "across ranks, replicated decode); world = tp*cp")
    return ap.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_cuda = torch.cuda.is_available()
    device = args.device or (f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(local_rank)

    dist = None
    comm = None
    tp = max(1, args.tp)
    cp = max(1, args.cp)
    if world > 1:
        import torch.distributed as tdist

        dist = tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")
        if tp > 1 or cp > 1:
            assert world == tp * cp, "bench --tp/--cp require world == tp*cp"
            from gpustack_amd.parallel import init_parallel

            comm = init_parallel(tp, 1, rank, cp_size=cp)

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    model = args.model
    cfg = EngineConfig(
        model=model,
        device=device,
        max_model_len=args.max_model_len,
        max_num_seqs=max(args.concurrency, 8),
        seed=0,
        kv_cache_dtype=args.kv_dtype,
        quantize_runtime=args.quantize,
        enable_prefix_caching=args.prefix_caching,
        speculative=({"method": args.speculative,
                      "num_draft_tokens": args.draft_tokens,
                      **({"draft_dir": args.draft_dir}
                         if args.draft_dir else {})}
                     if args.speculative else None),
        tp_size=tp if comm else 1,
        tp_rank=comm.tp_rank if comm else 0,
    )
    if not use_cuda:  # CPU smoke path: shrink everything
        cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=256, max_model_len=512,
                           tp_size=tp if comm else 1,
                           tp_rank=comm.tp_rank if comm else 0)
        args.concurrency = min(args.concurrency, 8)
        args.isl, args.osl = 32, 16
        model = "tiny"
    eng = LLMEngine(cfg, comm)
    is_driver = comm is None or rank == 0

    rng = random.Random(1234 + rank)
    vocab = cfg.spec.vocab_size
    params = SamplingParams(max_tokens=args.osl, ignore_eos=True)

    def refill():
        if not is_driver:
            return
        target = args.concurrency
        known = eng.scheduler.num_unfinished + len(eng._pending_ops)
        while known < target:
            toks = [rng.randrange(2, vocab) for _ in range(args.isl)]
            eng.add_request(toks, params)
            known += 1

    finished_count = [0]
    # TTFT sampling: with identical ISL/OSL the closed loop is periodic
    # (whole admission waves finish together every ~OSL steps), so a short
    # timed window can miss all admissions. Samples are therefore collected
    # from churn start onward — warmup tail AND timed region — which covers
    # at least one full admission cycle of steady-state TTFTs.
    ttft_track = [False]
    first_token_seen: set[str] = set()
    ttfts: list[float] = []

    def record_first_tokens(outs) -> None:
        if not ttft_track[0]:
            return
        for o in outs:
            if o.request_id in first_token_seen:
                continue
            first_token_seen.add(o.request_id)
            seq = eng.seqs.get(o.request_id)
            if seq is not None and seq.ttft is not None:
                ttfts.append(seq.ttft)

    def one_step() -> int:
        refill()
        outs = eng.step()
        finished_count[0] += sum(1 for o in outs if o.finished)
        record_first_tokens(outs)
        return len(outs)

    def barrier_sync():
        if use_cuda:
            torch.cuda.synchronize()
        if dist:
            dist.barrier()
            if use_cuda:
                torch.cuda.synchronize()

    # ---- warmup: run AT LEAST --warmup steps, then keep warming until the
    # continuous batcher is saturated so the timed region measures steady-
    # state serving regardless of the driver's step count (a small --steps
    # with a small --warmup would otherwise time the prefill admission ramp,
    # not the serving throughput the metric names). Saturation = >=90% of the
    # target concurrency running AND the last steps are decode-dominant
    # (each step emits roughly one token per running sequence).
    warmup_steps = 0
    for _ in range(args.warmup):
        one_step()
        warmup_steps += 1
    target = args.concurrency
    saturated_streak = 0
    warmup_deadline = time.perf_counter() + 300.0
    extra_cap = 2048
    done = False
    while not done:
        emitted = one_step()
        warmup_steps += 1
        extra_cap -= 1
        running = eng.num_running
        if running >= 0.9 * target and emitted >= 0.8 * max(running, 1):
            saturated_streak += 1
        else:
            saturated_streak = 0
        # churn equilibrium: the first admission wave must have cycled out
        # (>= concurrency finishes) so the timed window sees the true serving
        # mix of decode steps + refill prefills
        churned = finished_count[0] >= target
        if churned and not ttft_track[0]:
            ttft_track[0] = True
            for rid, s in eng.seqs.items():
                if s.first_token_time is not None:
                    first_token_seen.add(rid)
        done = ((saturated_streak >= 3 and churned) or extra_cap <= 0
                or time.perf_counter() > warmup_deadline)
        if dist is not None:
            # every rank must exit the same iteration (collectives inside
            # one_step stay aligned): exit only when ALL ranks are done
            dev = device if use_cuda else "cpu"
            flag = torch.tensor([1.0 if done else 0.0], device=dev)
            dist.all_reduce(flag, op=dist.ReduceOp.MIN)
            done = flag.item() >= 1.0
    barrier_sync()

    # ---- timed region: exactly K steps ----
    ttft_track[0] = True
    t0 = time.perf_counter()
    tokens = 0
    for _ in range(args.steps):
        refill()
        outs = eng.step()
        tokens += len(outs)
        record_first_tokens(outs)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # aggregate across ranks: total tokens, max elapsed
    if dist:
        dev = device if use_cuda else "cpu"
        te = torch.tensor([elapsed], device=dev)
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())
        if comm is not None:
            tot_tokens = tokens  # TP: one replica, every rank saw the batch
        else:
            tt = torch.tensor([float(tokens)], device=dev)
            dist.all_reduce(tt)
            tot_tokens = int(tt.item())
    else:
        tot_tokens = tokens

    ttfts.sort()
    p50_ttft_ms = (ttfts[len(ttfts) // 2] * 1000) if ttfts else None

    value = tot_tokens / elapsed if elapsed > 0 else 0.0
    if rank == 0:
        print(json.dumps({
            "metric": "output tokens/sec (serving, continuous batching)",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": world if world > 1 else args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "strong" if comm is not None else "weak",
            "vs_baseline": None,
            "dtype": (f"{cfg.dtype}+w4-weights" if args.quantize == "w4"
                      else cfg.dtype) if use_cuda else "float32-cpu-smoke",
            "data": "synthetic random-token prompts, random-init weights",
            "config": {
                "model": model,
                "global_batch": args.concurrency * (1 if comm is not None else (world if world > 1 else 1)),
                "seq_len": args.isl + args.osl,
                "parallelism": (f"tp{world}" if comm is not None
                                else f"dp{world if world > 1 else args.gpus}"),
                "isl": args.isl,
                "osl": args.osl,
                "concurrency_per_gpu": args.concurrency,
                "p50_ttft_ms": round(p50_ttft_ms, 2) if p50_ttft_ms else None,
                "timed_output_tokens": tot_tokens,
                "warmup_steps_actual": warmup_steps,
                "timed_region": "steady-state (warmup self-extends until "
                                "the batcher saturates, then exactly "
                                f"{args.steps} steps are timed)",
            },
        }))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
