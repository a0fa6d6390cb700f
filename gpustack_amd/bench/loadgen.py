"""First-party async load generator (replaces the reference's
benchmark-runner container; vllm-bench-serve-style).

Drives an OpenAI-compatible endpoint with synthetic prompts at either a
fixed request rate (Poisson arrivals — the BASELINE.json "fixed QPS" mode)
or fixed concurrency (closed loop), measuring per-request TTFT, TPOT and
token throughput with percentile aggregation (reference SLA metric names:
gpustack/schemas/benchmark.py:104-197).
"""
from __future__ import annotations

import asyncio
import json
import random
import time
from dataclasses import dataclass, field


@dataclass
class LoadSpec:
    mode: str = "concurrency"          # "concurrency" | "qps"
    value: float = 8                   # target concurrency or requests/sec
    duration_s: float = 30.0
    isl: int = 128                     # synthetic prompt tokens (approx)
    osl: int = 64                      # max output tokens
    model: str = ""
    stream: bool = True
    seed: int = 42
    api_key: str | None = None


@dataclass
class RequestResult:
    ok: bool = False
    ttft: float | None = None
    latency: float = 0.0
    output_tokens: int = 0
    prompt_tokens: int = 0
    itls: list[float] = field(default_factory=list)  # inter-token latencies
    error: str = ""


def _pct(vals: list[float], p: float) -> float | None:
    if not vals:
        return None
    vals = sorted(vals)
    idx = min(len(vals) - 1, int(p / 100 * len(vals)))
    return vals[idx]


class LoadGenerator:
    def __init__(self, base_url: str, spec: LoadSpec):
        self.base_url = base_url.rstrip("/")
        self.spec = spec
        self.results: list[RequestResult] = []
        self.rng = random.Random(spec.seed)

    def _prompt(self) -> str:
        # ~1 token per word for the byte tokenizer; content irrelevant
        words = [str(self.rng.randrange(10, 99)) for _ in range(self.spec.isl // 3)]
        return " ".join(words)

    async def _one_request(self, client) -> RequestResult:
        import httpx

        spec = self.spec
        res = RequestResult()
        body = {
            "model": spec.model,
            "messages": [{"role": "user", "content": self._prompt()}],
            "max_tokens": spec.osl,
            "ignore_eos": True,
            "stream": spec.stream,
        }
        headers = {}
        if spec.api_key:
            headers["Authorization"] = f"Bearer {spec.api_key}"
        t0 = time.perf_counter()
        try:
            if spec.stream:
                last_t = t0
                async with client.stream(
                    "POST", f"{self.base_url}/v1/chat/completions",
                    json=body, headers=headers,
                ) as resp:
                    if resp.status_code != 200:
                        res.error = f"http {resp.status_code}"
                        return res
                    async for line in resp.aiter_lines():
                        if not line.startswith("data:"):
                            continue
                        frag = line[5:].strip()
                        if frag == "[DONE]":
                            break
                        now = time.perf_counter()
                        try:
                            payload = json.loads(frag)
                        except json.JSONDecodeError:
                            continue
                        if payload.get("usage"):
                            res.output_tokens = payload["usage"]["completion_tokens"]
                            res.prompt_tokens = payload["usage"]["prompt_tokens"]
                        delta = payload.get("choices", [{}])[0].get("delta", {})
                        if delta.get("content"):
                            if res.ttft is None:
                                res.ttft = now - t0
                            else:
                                res.itls.append(now - last_t)
                            last_t = now
                res.ok = True
            else:
                r = await client.post(f"{self.base_url}/v1/chat/completions",
                                      json=body, headers=headers)
                res.ok = r.status_code == 200
                if res.ok:
                    u = r.json().get("usage", {})
                    res.output_tokens = u.get("completion_tokens", 0)
                    res.prompt_tokens = u.get("prompt_tokens", 0)
                    res.ttft = time.perf_counter() - t0
        except httpx.HTTPError as e:
            res.error = str(e)
        res.latency = time.perf_counter() - t0
        return res

    async def run(self) -> dict:
        import httpx

        spec = self.spec
        t_end = time.perf_counter() + spec.duration_s
        async with httpx.AsyncClient(timeout=httpx.Timeout(30.0, read=None)) as client:
            tasks: set[asyncio.Task] = set()

            async def tracked():
                r = await self._one_request(client)
                self.results.append(r)

            if spec.mode == "concurrency":
                sem = asyncio.Semaphore(int(spec.value))

                async def worker():
                    while time.perf_counter() < t_end:
                        async with sem:
                            await tracked()

                tasks = {asyncio.create_task(worker()) for _ in range(int(spec.value))}
                await asyncio.wait(tasks)
            else:  # fixed QPS, Poisson arrivals
                t0 = time.perf_counter()
                while time.perf_counter() < t_end:
                    tasks.add(asyncio.create_task(tracked()))
                    await asyncio.sleep(self.rng.expovariate(spec.value))
                    tasks = {t for t in tasks if not t.done()}
                if tasks:
                    await asyncio.wait(tasks, timeout=60)
        return self.summary()

    def raw_records(self, cap: int = 1000) -> list[dict]:
        """Per-request artifact rows (reference: benchmark artifacts,
        worker/benchmark/artifacts.py): capped to keep the DB row small."""
        out = []
        for r in self.results[:cap]:
            out.append({
                "ok": r.ok,
                "ttft_ms": round(r.ttft * 1000, 2) if r.ttft is not None else None,
                "latency_ms": round(r.latency * 1000, 2),
                "output_tokens": r.output_tokens,
                "prompt_tokens": r.prompt_tokens,
                "error": r.error or None,
            })
        return out

    def summary(self) -> dict:
        ok = [r for r in self.results if r.ok]
        dur = self.spec.duration_s
        ttfts = [r.ttft for r in ok if r.ttft is not None]
        itls = [x for r in ok for x in r.itls]
        out_tokens = sum(r.output_tokens for r in ok)
        in_tokens = sum(r.prompt_tokens for r in ok)
        return {
            "requests": len(self.results),
            "successful_requests": len(ok),
            "failed_requests": len(self.results) - len(ok),
            "duration_s": round(dur, 3),
            "request_rate": round(len(ok) / dur, 3),
            "output_tokens": out_tokens,
            "prompt_tokens": in_tokens,
            "output_tps": round(out_tokens / dur, 2),
            "total_tps": round((out_tokens + in_tokens) / dur, 2),
            "ttft_p50_ms": round(_pct(ttfts, 50) * 1000, 2) if ttfts else None,
            "ttft_p90_ms": round(_pct(ttfts, 90) * 1000, 2) if ttfts else None,
            "ttft_p99_ms": round(_pct(ttfts, 99) * 1000, 2) if ttfts else None,
            "ttft_mean_ms": round(sum(ttfts) / len(ttfts) * 1000, 2) if ttfts else None,
            "tpot_p50_ms": round(_pct(itls, 50) * 1000, 2) if itls else None,
            "tpot_p99_ms": round(_pct(itls, 99) * 1000, 2) if itls else None,
            "tpot_mean_ms": round(sum(itls) / len(itls) * 1000, 2) if itls else None,
        }


def run_load(base_url: str, spec: LoadSpec,
             with_artifacts: bool = False) -> dict:
    gen = LoadGenerator(base_url, spec)
    res = asyncio.run(gen.run())
    if with_artifacts:
        res["artifacts"] = gen.raw_records()
    return res
