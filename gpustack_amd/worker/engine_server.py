"""Per-instance OpenAI-compatible engine server.

The first-party replacement for the vLLM/SGLang containers the reference
launches: the worker's serve manager starts this as a
subprocess per ModelInstance (`python -m gpustack_amd.worker.engine_server`)
with HIP_VISIBLE_DEVICES pinned to the scheduled GPUs.

A dedicated engine thread runs the continuous-batching loop; request
handlers feed prompts in and stream sampled tokens out through per-request
asyncio queues. Endpoints: /v1/chat/completions, /v1/completions,
/v1/models, /health, /metrics.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import logging
import threading
import time
import uuid

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

logger = logging.getLogger(__name__)


class ByteTokenizer:
    """Reversible fallback tokenizer for preset/random-init serving (no
    checkpoint, no vocab files): UTF-8 bytes offset past the special ids."""

    OFFSET = 16

    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size
        self.eos_token_id = 1

    def encode(self, text: str) -> list[int]:
        return [b % (self.vocab_size - self.OFFSET) + self.OFFSET
                for b in text.encode("utf-8")]

    def decode(self, ids: list[int]) -> str:
        return bytes((i - self.OFFSET) % 256 for i in ids if i >= self.OFFSET).decode(
            "utf-8", errors="replace"
        )

    def apply_chat_template(self, messages: list[dict]) -> str:
        parts = [f"{m.get('role', 'user')}: {m.get('content', '')}" for m in messages]
        return "\n".join(parts) + "\nassistant:"


def load_tokenizer(model_dir: str | None, vocab_size: int):
    if model_dir:
        try:
            from transformers import AutoTokenizer

            return AutoTokenizer.from_pretrained(model_dir)
        except Exception:  # noqa: BLE001
            logger.warning("falling back to ByteTokenizer for %s", model_dir)
    return ByteTokenizer(vocab_size)


class EngineRunner:
    """Owns the LLMEngine + step loop thread; bridges to asyncio."""

    def __init__(self, engine_cfg, served_name: str, comm=None):
        from ..engine import LLMEngine

        self.engine = LLMEngine(engine_cfg, comm)
        self.served_name = served_name
        self.tokenizer = load_tokenizer(engine_cfg.model_dir, engine_cfg.spec.vocab_size)
        self.loop: asyncio.AbstractEventLoop | None = None
        self._queues: dict[str, asyncio.Queue] = {}
        self._aux_jobs: "list[tuple]" = []  # (fn, args, asyncio.Future)
        self._lock = threading.Lock()
        self._wake = threading.Event()
        self._stop = False
        self.stats = {"requests": 0, "generated_tokens": 0, "prompt_tokens": 0}
        # latency histograms (vLLM-parity observability): TTFT per request
        # and inter-token gap per emitted token, fixed second buckets
        self.ttft_buckets = [0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0]
        self.tpot_buckets = [0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0]
        self.ttft_hist = [0] * (len(self.ttft_buckets) + 1)
        self.ttft_sum = 0.0
        self.ttft_count = 0
        self.tpot_hist = [0] * (len(self.tpot_buckets) + 1)
        self.tpot_sum = 0.0
        self.tpot_count = 0
        self._first_seen: dict[str, float] = {}  # rid -> last token time
        self.thread = threading.Thread(target=self._run, name="engine-loop", daemon=True)

    def start(self, loop: asyncio.AbstractEventLoop) -> None:
        self.loop = loop
        self.thread.start()

    def stop(self) -> None:
        self._stop = True
        self._wake.set()

    def _run_aux_jobs(self) -> None:
        with self._lock:
            jobs, self._aux_jobs = self._aux_jobs, []
        for fn, args, fut in jobs:
            try:
                result = fn(*args)
                self.loop.call_soon_threadsafe(fut.set_result, result)
            except Exception as e:  # noqa: BLE001
                self.loop.call_soon_threadsafe(fut.set_exception, e)

    async def run_aux(self, fn, *args):
        """Run fn on the engine thread (between steps) and await the result
        — used by non-generative endpoints like /v1/embeddings."""
        fut = asyncio.get_running_loop().create_future()
        with self._lock:
            self._aux_jobs.append((fn, args, fut))
        self._wake.set()
        return await fut

    def _run(self) -> None:
        tp = self.engine.comm.world_size > 1
        while not self._stop:
            if self._aux_jobs:
                self._run_aux_jobs()
            if tp:
                # coordinated loop: the tp_active broadcast doubles as the
                # idle heartbeat for follower ranks
                if not self.engine.tp_active():
                    self._wake.wait(timeout=0.02)
                    self._wake.clear()
                    continue
            elif not self.engine.has_unfinished():
                self._wake.wait(timeout=0.05)
                self._wake.clear()
                continue
            try:
                outputs = self.engine.step()
            except Exception as e:  # noqa: BLE001
                logger.exception("engine step failed")
                with self._lock:
                    for q in self._queues.values():
                        self._push(q, {"error": str(e), "finished": True})
                time.sleep(1)
                continue
            now = time.monotonic()
            for out in outputs:
                with self._lock:
                    q = self._queues.get(out.request_id)
                if q is not None:
                    self.stats["generated_tokens"] += 1
                    prev = self._first_seen.get(out.request_id)
                    if prev is None:
                        seq = self.engine.seqs.get(out.request_id)
                        if seq is not None and seq.first_token_time is not None:
                            self._observe(
                                "ttft",
                                seq.first_token_time - seq.arrival_time)
                    else:
                        self._observe("tpot", now - prev)
                    if out.finished:
                        self._first_seen.pop(out.request_id, None)
                    else:
                        self._first_seen[out.request_id] = now
                    self._push(q, {
                        "token_id": out.token_id,
                        "finished": out.finished,
                        "finish_reason": out.finish_reason,
                        "logprob": out.logprob,
                        "top_logprobs": out.top_logprobs,
                    })

    def _observe(self, kind: str, v: float) -> None:
        buckets = self.ttft_buckets if kind == "ttft" else self.tpot_buckets
        hist = self.ttft_hist if kind == "ttft" else self.tpot_hist
        for i, b in enumerate(buckets):
            if v <= b:
                hist[i] += 1
                break
        else:
            hist[-1] += 1
        if kind == "ttft":
            self.ttft_sum += v
            self.ttft_count += 1
        else:
            self.tpot_sum += v
            self.tpot_count += 1

    def _push(self, q: asyncio.Queue, item: dict) -> None:
        assert self.loop is not None
        self.loop.call_soon_threadsafe(q.put_nowait, item)

    def submit(self, token_ids: list[int], params) -> tuple[str, asyncio.Queue]:
        rid = f"req-{uuid.uuid4().hex[:16]}"
        q: asyncio.Queue = asyncio.Queue()
        with self._lock:
            self._queues[rid] = q
        self.engine.add_request(token_ids, params, request_id=rid)
        self.stats["requests"] += 1
        self.stats["prompt_tokens"] += len(token_ids)
        self._wake.set()
        return rid, q

    def ensure_token_table(self) -> None:
        """Build the id->string table guided-JSON decoding probes (lazy:
        one decode pass over the vocab, first guided request only).

        Strings are derived as decode([anchor, i]) minus decode([anchor])
        rather than decode([i]): for SentencePiece / byte-level-BPE
        tokenizers a lone-token decode strips the leading '▁'/space, so the
        characters the guided machine validates would diverge from the final
        decode(tokens) text and break the valid-JSON/regex guarantee. The
        anchor context makes each token's string match what it contributes
        mid-sequence. (Multi-token UTF-8 sequences still decode to U+FFFD
        per partial token — an inherent limit of a static per-id table;
        guided grammars constrain ASCII structure, where the diff is exact.)
        """
        if getattr(self, "_token_table_done", False):
            return
        self._token_table_done = True
        tok = self.tokenizer
        anchor = None
        prefix = ""
        try:
            ids = tok.encode("x")
            ids = [i for i in ids] if not hasattr(ids, "ids") else list(ids.ids)
            if ids:
                anchor = ids[-1]
                prefix = tok.decode([anchor])
        except Exception:  # noqa: BLE001
            anchor = None
        table = []
        for i in range(self.engine.cfg.spec.vocab_size):
            try:
                if anchor is not None:
                    s = tok.decode([anchor, i])
                    table.append(s[len(prefix):] if s.startswith(prefix)
                                 else tok.decode([i]))
                else:
                    table.append(tok.decode([i]))
            except Exception:  # noqa: BLE001
                table.append("")
        self.engine.set_token_table(table)

    def release(self, rid: str) -> None:
        with self._lock:
            self._queues.pop(rid, None)

    def abort(self, rid: str) -> None:
        self.engine.abort_request(rid)
        self._first_seen.pop(rid, None)
        self.release(rid)


def _tool_call_message(rid: str, body: dict, text: str) -> dict:
    """Shape a generated (guided) JSON into an OpenAI tool_calls message."""
    if body.get("_tool_envelope"):
        try:
            doc = json.loads(text)
            name = doc.get("name")
            args = json.dumps(doc.get("arguments", {}))
        except json.JSONDecodeError:
            name, args = None, text
    else:
        name, args = body.get("_tool_name"), text
    return {"role": "assistant", "content": None,
            "tool_calls": [{"id": f"call_{rid}", "type": "function",
                            "function": {"name": name, "arguments": args}}]}


def _stop_strings(body: dict) -> list[str]:
    stop = body.get("stop")
    if stop is None:
        return []
    if isinstance(stop, str):
        return [stop]
    return [s for s in stop if isinstance(s, str)]


def _sampling_params(body: dict, eos_token_id: int, tokenizer=None):
    from ..engine import SamplingParams

    mt = body.get("max_tokens") or body.get("max_completion_tokens") or 256
    # guided JSON: explicit guided_json (bool | schema) or the OpenAI
    # response_format surface (json_object / json_schema)
    gj = body.get("guided_json")
    rf = body.get("response_format") or {}
    if gj is None and rf.get("type") == "json_object":
        gj = True
    elif gj is None and rf.get("type") == "json_schema":
        gj = (rf.get("json_schema") or {}).get("schema") or True
    guided = None
    if body.get("guided_choice") and tokenizer is not None:
        seqs = []
        for choice in body["guided_choice"]:
            ids = tokenizer.encode(choice)
            if hasattr(ids, "ids"):
                ids = ids.ids
            seqs.append(tuple(ids))
        guided = tuple(seqs)
    sp = SamplingParams(
        temperature=float(body.get("temperature", 1.0) or 0.0)
        if body.get("temperature") is not None else 1.0,
        top_p=float(body.get("top_p", 1.0)),
        top_k=int(body.get("top_k", 0)),
        min_p=float(body.get("min_p", 0.0)),
        presence_penalty=float(body.get("presence_penalty", 0.0)),
        frequency_penalty=float(body.get("frequency_penalty", 0.0)),
        repetition_penalty=float(body.get("repetition_penalty", 1.0)),
        logit_bias=body.get("logit_bias"),
        max_tokens=int(mt),
        min_tokens=int(body.get("min_tokens") or 0),
        priority=int(body.get("priority") or 0),
        ignore_eos=bool(body.get("ignore_eos", False)),
        seed=body.get("seed"),
        logprobs=bool(body.get("logprobs")) or int(body.get("top_logprobs") or 0) > 0,
        top_logprobs=int(body.get("top_logprobs") or 0),
        guided_token_seqs=guided,
        guided_json=gj,
        guided_regex=body.get("guided_regex"),
        guided_grammar=body.get("guided_grammar"),
        eos_token_id=eos_token_id,
    )
    if gj is not None or sp.guided_regex is not None \
            or sp.guided_grammar is not None:
        sp.ignore_eos = False
    if guided:
        sp.ignore_eos = False
        sp.max_tokens = max(len(c) for c in guided) + 1
    return sp


def _apply_lora_routing(sp, body: dict, runner: "EngineRunner") -> None:
    """Per-request adapter resolution (reference: per-LoRA child model
    routes, gpustack/server/lora_model_routes.py): requesting a live
    adapter's name as `model` — or an explicit `lora_name` — serves the
    base model with that adapter applied to this request's rows."""
    name = body.get("lora_name") or body.get("model")
    if name and name != runner.served_name \
            and name in runner.engine.lora_names():
        sp.lora_name = name


def create_encoder_app(model_dir: str, served_name: str,
                       device: str = "cpu") -> FastAPI:
    """Serving app for cross-encoder reranker checkpoints
    (models/encoder.py): /v1/rerank scores (query, doc) PAIRS through the
    bidirectional encoder — the real reranker path; generative endpoints
    404 (the reference's reranker-category instances behave the same)."""
    import torch as _torch

    from ..models.encoder import (CrossEncoderModel, CrossEncoderRunner,
                                  EncoderSpec)

    spec = EncoderSpec.from_dir(model_dir)
    dtype = _torch.bfloat16 if device.startswith("cuda") else _torch.float32
    model = CrossEncoderModel(spec, device=device, dtype=dtype)
    from pathlib import Path as _Path

    if list(_Path(model_dir).glob("*.safetensors")):
        model.load_dir(model_dir)
    else:
        model.random_init(seed=0)
    tok = load_tokenizer(model_dir, spec.vocab_size)

    def _special(name: str, fallback: int) -> int:
        getter = getattr(tok, "token_to_id", None)
        if getter is not None:
            for cand in name.split("|"):
                tid = getter(cand)
                if tid is not None:
                    return tid
        return fallback
    if spec.is_roberta:
        cls_id = _special("<s>|[CLS]", 0)
        sep_id = _special("</s>|[SEP]", 2)
    else:
        cls_id = _special("[CLS]|<s>", min(101, spec.vocab_size - 2))
        sep_id = _special("[SEP]|</s>", min(102, spec.vocab_size - 1))
    ce = CrossEncoderRunner(model, cls_id, sep_id, device=device)
    stats = {"requests": 0, "pairs": 0}
    app = FastAPI(title=f"gpustack_amd reranker: {served_name}")

    def _enc(text: str) -> list[int]:
        ids = tok.encode(text)
        if hasattr(ids, "ids"):
            ids = ids.ids
        return list(ids) or [0]

    @app.get("/health")
    async def health():
        return {"status": "ok", "model": served_name, "mode": "reranker"}

    @app.get("/v1/models")
    async def models():
        return {"object": "list", "data": [{
            "id": served_name, "object": "model", "owned_by": "gpustack_amd",
            "meta": {"category": "reranker"}}]}

    @app.get("/metrics")
    async def metrics():
        return Response("\n".join([
            "# TYPE gpustack_engine_requests_total counter",
            f"gpustack_engine_requests_total {stats['requests']}",
            "# TYPE gpustack_rerank_pairs_total counter",
            f"gpustack_rerank_pairs_total {stats['pairs']}",
        ]) + "\n", media_type="text/plain")

    @app.post("/v1/rerank")
    @app.post("/rerank")
    async def rerank(request: Request):
        import math as _math

        body = await request.json()
        query = body.get("query", "")
        docs = body.get("documents", [])
        top_n = int(body.get("top_n", len(docs)) or len(docs))
        stats["requests"] += 1
        stats["pairs"] += len(docs)
        q_ids = _enc(query)
        doc_ids = [_enc(d) for d in docs]
        logits = await asyncio.to_thread(ce.score, q_ids, doc_ids)
        scored = [{"index": i,
                   "relevance_score": 1.0 / (1.0 + _math.exp(-x)),
                   "document": {"text": docs[i]}}
                  for i, x in enumerate(logits)]
        scored.sort(key=lambda r: -r["relevance_score"])
        ntok = len(q_ids) * len(docs) + sum(len(d) for d in doc_ids)
        return {"model": served_name, "results": scored[:top_n],
                "usage": {"prompt_tokens": ntok, "total_tokens": ntok}}

    @app.post("/v1/score")
    async def score(request: Request):
        """Pairwise relevance (sigmoid of the pair logit)."""
        import math as _math

        body = await request.json()
        t1 = body.get("text_1", "")
        t2 = body.get("text_2", [])
        if isinstance(t2, str):
            t2 = [t2]
        stats["requests"] += 1
        stats["pairs"] += len(t2)
        logits = await asyncio.to_thread(
            ce.score, _enc(t1), [_enc(t) for t in t2])
        return {"object": "list", "model": served_name,
                "data": [{"index": i,
                          "score": 1.0 / (1.0 + _math.exp(-x))}
                         for i, x in enumerate(logits)]}

    return app


def create_app(runner: EngineRunner) -> FastAPI:
    app = FastAPI(title="gpustack_amd-engine")

    @app.on_event("startup")
    async def _startup():
        runner.start(asyncio.get_running_loop())

    @app.on_event("shutdown")
    async def _shutdown():
        # graceful drain: stop the engine loop and fail pending requests
        # cleanly instead of cutting streams mid-token (uvicorn delivers
        # this on SIGTERM from the serve manager)
        runner.stop()
        with runner._lock:
            queues = list(runner._queues.values())
        for q in queues:
            runner._push(q, {"error": "server shutting down",
                             "finished": True})

    @app.get("/health")
    async def health():
        return {"status": "ok", "model": runner.served_name}

    @app.get("/v1/models")
    async def models():
        data = [{"id": runner.served_name, "object": "model",
                 "owned_by": "gpustack_amd"}]
        data.extend({"id": n, "object": "model", "owned_by": "gpustack_amd",
                     "parent": runner.served_name}
                    for n in runner.engine.lora_names())
        return {"object": "list", "data": data}

    # vLLM-compatible dynamic adapter management (reference: gpustack mounts
    # per-LoRA child routes over these, server/lora_model_routes.py)
    @app.post("/v1/load_lora_adapter")
    async def load_lora(request: Request):
        body = await request.json()
        name, path = body.get("lora_name"), body.get("lora_path")
        if not name or not path:
            raise HTTPException(400, "lora_name and lora_path are required")
        if name == runner.served_name:
            raise HTTPException(400, "adapter name collides with served model")
        try:
            # on the engine thread: the bank mutates between steps only
            await runner.run_aux(runner.engine.add_lora, name, path)
        except (FileNotFoundError, ValueError) as e:
            raise HTTPException(400, str(e))
        return {"status": "ok", "lora_name": name}

    @app.post("/v1/unload_lora_adapter")
    async def unload_lora(request: Request):
        body = await request.json()
        name = body.get("lora_name")
        if not name:
            raise HTTPException(400, "lora_name is required")
        ok = await runner.run_aux(runner.engine.remove_lora, name)
        if not ok:
            raise HTTPException(404, f"no adapter named {name!r}")
        return {"status": "ok", "lora_name": name}

    @app.get("/metrics")
    async def metrics():
        e = runner.engine
        lines = [
            "# TYPE gpustack_engine_requests_total counter",
            f"gpustack_engine_requests_total {runner.stats['requests']}",
            "# TYPE gpustack_engine_generated_tokens_total counter",
            f"gpustack_engine_generated_tokens_total {runner.stats['generated_tokens']}",
            "# TYPE gpustack_engine_prompt_tokens_total counter",
            f"gpustack_engine_prompt_tokens_total {runner.stats['prompt_tokens']}",
            "# TYPE gpustack_engine_num_running gauge",
            f"gpustack_engine_num_running {e.num_running}",
            "# TYPE gpustack_engine_num_waiting gauge",
            f"gpustack_engine_num_waiting {e.num_waiting}",
            "# TYPE gpustack_engine_kv_blocks_free gauge",
            f"gpustack_engine_kv_blocks_free {e.scheduler.kv.allocator.num_free}",
            "# TYPE gpustack_engine_kv_blocks_total gauge",
            f"gpustack_engine_kv_blocks_total {e.scheduler.kv.allocator.num_blocks}",
        ]
        for kind, buckets, hist, hsum, hcount in (
            ("ttft", runner.ttft_buckets, runner.ttft_hist,
             runner.ttft_sum, runner.ttft_count),
            ("time_per_output_token", runner.tpot_buckets,
             runner.tpot_hist, runner.tpot_sum, runner.tpot_count),
        ):
            name = f"gpustack_engine_{kind}_seconds"
            lines.append(f"# TYPE {name} histogram")
            cum = 0
            for b, n in zip(buckets, hist):
                cum += n
                lines.append(f'{name}_bucket{{le="{b}"}} {cum}')
            cum += hist[-1]
            lines.append(f'{name}_bucket{{le="+Inf"}} {cum}')
            lines.append(f"{name}_sum {hsum}")
            lines.append(f"{name}_count {hcount}")
        return Response("\n".join(lines) + "\n", media_type="text/plain; version=0.0.4")

    async def _generate(request: Request, body: dict, prompt_ids: list[int],
                        kind: str, echo_text_prefix: str = ""):
        n = max(1, int(body.get("n", 1) or 1))
        if n > 1:
            return await _generate_n(request, body, prompt_ids, kind, n,
                                     echo_text_prefix)
        params = _sampling_params(body, runner.engine.cfg.spec.eos_token_id,
                                  runner.tokenizer)
        _apply_lora_routing(params, body, runner)
        if params.guided_json is not None or params.guided_regex is not None \
                or params.guided_grammar is not None:
            runner.ensure_token_table()
        stop_strs = _stop_strings(body)
        rid, q = runner.submit(prompt_ids, params)
        created = int(time.time())
        stream = bool(body.get("stream"))
        model_name = runner.served_name

        def chat_chunk(delta: dict, finish: str | None):
            return {
                "id": rid, "object": "chat.completion.chunk", "created": created,
                "model": model_name,
                "choices": [{"index": 0, "delta": delta, "finish_reason": finish}],
            }

        def text_chunk(text: str, finish: str | None):
            return {
                "id": rid, "object": "text_completion", "created": created,
                "model": model_name,
                "choices": [{"index": 0, "text": text, "finish_reason": finish}],
            }

        tool_mode = kind == "chat" and bool(body.get("_tool_name")
                                            or body.get("_tool_envelope"))
        if stream:
            async def gen():
                tokens: list[int] = []
                sent_len = 0
                pending_lps: list[float] = []  # logprobs since the last emit
                try:
                    if kind == "chat":
                        yield f"data: {json.dumps(chat_chunk({'role': 'assistant'}, None))}\n\n"
                    while True:
                        if await request.is_disconnected():
                            runner.abort(rid)
                            return
                        item = await q.get()
                        if "error" in item:
                            yield f"data: {json.dumps({'error': {'message': item['error']}})}\n\n"
                            break
                        tokens.append(item["token_id"])
                        if item.get("logprob") is not None:
                            pending_lps.append(item["logprob"])
                        text = runner.tokenizer.decode(tokens)
                        if stop_strs:  # stop strings apply mid-stream too
                            cuts = [text.find(ss) for ss in stop_strs
                                    if ss in text]
                            if cuts:
                                cut = min(cuts)
                                runner.abort(rid)
                                tail = text[sent_len:cut]
                                if tail and not tool_mode:
                                    payload = (chat_chunk({"content": tail}, None)
                                               if kind == "chat"
                                               else text_chunk(tail, None))
                                    yield f"data: {json.dumps(payload)}\n\n"
                                usage = {
                                    "prompt_tokens": len(prompt_ids),
                                    "completion_tokens": len(tokens),
                                    "total_tokens": len(prompt_ids) + len(tokens),
                                }
                                payload = (chat_chunk({}, "stop")
                                           if kind == "chat"
                                           else text_chunk("", "stop"))
                                payload["usage"] = usage
                                yield f"data: {json.dumps(payload)}\n\n"
                                yield "data: [DONE]\n\n"
                                break
                        new = text[sent_len:]
                        # hold back partial unicode replacement chars
                        if new and not new.endswith("�") and not tool_mode:
                            sent_len = len(text)
                            payload = (chat_chunk({"content": new}, None)
                                       if kind == "chat" else text_chunk(new, None))
                            if body.get("logprobs") and pending_lps:
                                if kind == "chat":
                                    payload["choices"][0]["logprobs"] = {
                                        "content": [{"token": new,
                                                     "logprob": lp}
                                                    for lp in pending_lps]}
                                else:
                                    payload["choices"][0]["logprobs"] = {
                                        "tokens": [new] + [""] * (len(pending_lps) - 1),
                                        "token_logprobs": list(pending_lps)}
                                pending_lps.clear()
                            yield f"data: {json.dumps(payload)}\n\n"
                        if item["finished"] and tool_mode:
                            msg = _tool_call_message(rid, body,
                                                     runner.tokenizer.decode(tokens))
                            yield f"data: {json.dumps(chat_chunk({'tool_calls': msg['tool_calls']}, None))}\n\n"
                            usage = {
                                "prompt_tokens": len(prompt_ids),
                                "completion_tokens": len(tokens),
                                "total_tokens": len(prompt_ids) + len(tokens),
                            }
                            payload = chat_chunk({}, "tool_calls")
                            payload["usage"] = usage
                            yield f"data: {json.dumps(payload)}\n\n"
                            yield "data: [DONE]\n\n"
                            break
                        if item["finished"]:
                            fin = item.get("finish_reason") or "stop"
                            usage = {
                                "prompt_tokens": len(prompt_ids),
                                "completion_tokens": len(tokens),
                                "total_tokens": len(prompt_ids) + len(tokens),
                            }
                            payload = (chat_chunk({}, fin) if kind == "chat"
                                       else text_chunk("", fin))
                            payload["usage"] = usage
                            yield f"data: {json.dumps(payload)}\n\n"
                            yield "data: [DONE]\n\n"
                            break
                finally:
                    runner.release(rid)

            return StreamingResponse(gen(), media_type="text/event-stream")

        tokens: list[int] = []
        lps: list[float | None] = []
        tops: list[list | None] = []
        finish = "stop"
        try:
            while True:
                item = await q.get()
                if "error" in item:
                    raise HTTPException(500, item["error"])
                tokens.append(item["token_id"])
                lps.append(item.get("logprob"))
                tops.append(item.get("top_logprobs"))
                if stop_strs:
                    t = runner.tokenizer.decode(tokens)
                    if any(ss in t for ss in stop_strs):
                        runner.abort(rid)
                        finish = "stop"
                        break
                if item["finished"]:
                    finish = item.get("finish_reason") or "stop"
                    break
        finally:
            runner.release(rid)
        text = runner.tokenizer.decode(tokens)
        if stop_strs:
            cuts = [text.find(ss) for ss in stop_strs if ss in text]
            if cuts:
                text = text[:min(cuts)]
        usage = {
            "prompt_tokens": len(prompt_ids),
            "completion_tokens": len(tokens),
            "total_tokens": len(prompt_ids) + len(tokens),
        }
        want_lp = (bool(body.get("logprobs"))
                   or int(body.get("top_logprobs") or 0) > 0) \
            and any(x is not None for x in lps)
        if kind == "chat" and (body.get("_tool_name") or body.get("_tool_envelope")):
            return JSONResponse({
                "id": rid, "object": "chat.completion", "created": created,
                "model": model_name,
                "choices": [{"index": 0,
                             "message": _tool_call_message(rid, body, text),
                             "finish_reason": "tool_calls"}],
                "usage": usage,
            })
        if kind == "chat":
            choice = {"index": 0,
                      "message": {"role": "assistant", "content": text},
                      "finish_reason": finish}
            if want_lp:
                choice["logprobs"] = {"content": [
                    {"token": runner.tokenizer.decode([t]), "logprob": lp,
                     **({"top_logprobs": [
                         {"token": runner.tokenizer.decode([tid]),
                          "logprob": tlp} for tid, tlp in top]}
                        if top else {})}
                    for t, lp, top in zip(tokens, lps, tops)
                ]}
            return JSONResponse({
                "id": rid, "object": "chat.completion", "created": created,
                "model": model_name, "choices": [choice], "usage": usage,
            })
        choice = {"index": 0, "text": echo_text_prefix + text,
                  "finish_reason": finish}
        if want_lp:
            choice["logprobs"] = {
                "tokens": [runner.tokenizer.decode([t]) for t in tokens],
                "token_logprobs": lps,
            }
        return JSONResponse({
            "id": rid, "object": "text_completion", "created": created,
            "model": model_name, "choices": [choice], "usage": usage,
        })

    def _setup_tool_calling(body: dict) -> None:
        """OpenAI tool calling (reference: engines' function-calling surface
        proxied by gpustack). A forced tool_choice or "required" constrains
        generation with the function's JSON-Schema via guided decoding;
        "auto" stays unconstrained (plain content answer)."""
        tools = body.get("tools") or []
        tc = body.get("tool_choice")
        funcs = {t["function"]["name"]: t["function"]
                 for t in tools if t.get("type") == "function"
                 and t.get("function", {}).get("name")}
        if not funcs or tc == "none":
            return
        if isinstance(tc, dict) and tc.get("type") == "function":
            name = (tc.get("function") or {}).get("name")
            if name not in funcs:
                raise HTTPException(400, f"unknown tool {name!r}")
            body["guided_json"] = funcs[name].get("parameters") or True
            body["_tool_name"] = name
        elif tc == "required":
            body["guided_json"] = {
                "type": "object",
                "properties": {
                    "name": {"enum": sorted(funcs)},
                    "arguments": {},  # any JSON value
                },
            }
            body["_tool_envelope"] = True

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        body = await request.json()
        messages = body.get("messages") or []
        _setup_tool_calling(body)
        tok = runner.tokenizer
        if hasattr(tok, "apply_chat_template"):
            try:
                prompt = tok.apply_chat_template(
                    messages, tokenize=False, add_generation_prompt=True,
                    tools=body.get("tools") or None)
            except TypeError:
                prompt = tok.apply_chat_template(messages)
        else:
            prompt = "\n".join(f"{m['role']}: {m['content']}" for m in messages)
        ids = tok.encode(prompt)
        if hasattr(ids, "ids"):
            ids = ids.ids
        return await _generate(request, body, list(ids), "chat")

    async def _generate_n(request: Request, body: dict, prompt_ids: list[int],
                          kind: str, n: int, echo_text_prefix: str = ""):
        """OpenAI `n` choices: n independent engine requests batched by the
        continuous-batching scheduler; seeded requests get seed+i so the
        choices differ."""
        import dataclasses

        base = _sampling_params(body, runner.engine.cfg.spec.eos_token_id,
                                runner.tokenizer)
        _apply_lora_routing(base, body, runner)
        if base.guided_json is not None or base.guided_regex is not None \
                or base.guided_grammar is not None:
            runner.ensure_token_table()
        stop_strs = _stop_strings(body)
        subs = []
        for i in range(n):
            p = dataclasses.replace(
                base, seed=(base.seed + i) if base.seed is not None else None)
            subs.append(runner.submit(prompt_ids, p))
        created = int(time.time())
        model_name = runner.served_name
        if body.get("stream"):
            async def gen():
                merged: asyncio.Queue = asyncio.Queue()

                async def pump(idx, q):
                    while True:
                        item = await q.get()
                        await merged.put((idx, item))
                        if item.get("finished") or "error" in item:
                            return

                tasks = [asyncio.ensure_future(pump(i, q))
                         for i, (_, q) in enumerate(subs)]
                toks = [[] for _ in range(n)]
                sent = [0] * n
                done = 0
                try:
                    while done < n:
                        if await request.is_disconnected():
                            for rid, _ in subs:
                                runner.abort(rid)
                            return
                        idx, item = await merged.get()
                        if "error" in item:
                            yield f"data: {json.dumps({'error': {'message': item['error']}})}\n\n"
                            done += 1
                            continue
                        toks[idx].append(item["token_id"])
                        text = runner.tokenizer.decode(toks[idx])
                        new = text[sent[idx]:]
                        fin = item.get("finished")
                        if new and not new.endswith("�"):
                            sent[idx] = len(text)
                            delta = ({"content": new} if kind == "chat"
                                     else None)
                            ch = {"index": idx,
                                  "finish_reason": (item.get("finish_reason")
                                                    or "stop") if fin else None}
                            if kind == "chat":
                                ch["delta"] = delta
                            else:
                                ch["text"] = new
                            payload = {"id": subs[0][0], "created": created,
                                       "model": model_name,
                                       "object": ("chat.completion.chunk"
                                                  if kind == "chat"
                                                  else "text_completion"),
                                       "choices": [ch]}
                            yield f"data: {json.dumps(payload)}\n\n"
                        if fin:
                            done += 1
                    yield "data: [DONE]\n\n"
                finally:
                    for t in tasks:
                        t.cancel()
                    for rid, _ in subs:
                        runner.release(rid)

            return StreamingResponse(gen(), media_type="text/event-stream")

        choices = []
        total_out = 0
        try:
            for i, (rid, q) in enumerate(subs):
                toks = []
                finish = "stop"
                while True:
                    item = await q.get()
                    if "error" in item:
                        raise HTTPException(500, item["error"])
                    toks.append(item["token_id"])
                    if stop_strs:
                        t = runner.tokenizer.decode(toks)
                        if any(ss in t for ss in stop_strs):
                            runner.abort(rid)
                            break
                    if item["finished"]:
                        finish = item.get("finish_reason") or "stop"
                        break
                text = runner.tokenizer.decode(toks)
                if stop_strs:
                    cuts = [text.find(ss) for ss in stop_strs if ss in text]
                    if cuts:
                        text = text[:min(cuts)]
                total_out += len(toks)
                if kind == "chat":
                    choices.append({"index": i,
                                    "message": {"role": "assistant",
                                                "content": text},
                                    "finish_reason": finish})
                else:
                    choices.append({"index": i,
                                    "text": echo_text_prefix + text,
                                    "finish_reason": finish})
        finally:
            for rid, _ in subs:
                runner.release(rid)
        usage = {"prompt_tokens": len(prompt_ids) * n,
                 "completion_tokens": total_out,
                 "total_tokens": len(prompt_ids) * n + total_out}
        return JSONResponse({
            "id": subs[0][0],
            "object": "chat.completion" if kind == "chat" else "text_completion",
            "created": created, "model": model_name,
            "choices": choices, "usage": usage,
        })

    @app.post("/v1/score")
    @app.post("/score")
    async def score(request: Request):
        """Sentence-pair scoring (reference routes /v1/score to engines that
        serve cross-encoder models; first-party design: embedding cosine
        from the same engine, matching /v1/rerank)."""
        body = await request.json()
        t1 = body.get("text_1") or body.get("query") or ""
        t2 = body.get("text_2") or body.get("documents") or []
        if isinstance(t2, str):
            t2 = [t2]
        tok = runner.tokenizer

        def enc(text):
            ids = tok.encode(text)
            if hasattr(ids, "ids"):
                ids = ids.ids
            return list(ids) or [0]

        prompts = [enc(t1)] + [enc(d) for d in t2]
        vecs = await runner.run_aux(runner.engine.runner.embed, prompts, "mean")
        import math

        def cos(a, b):
            num = sum(x * y for x, y in zip(a, b))
            den = math.sqrt(sum(x * x for x in a)) * math.sqrt(sum(y * y for y in b))
            return num / den if den else 0.0

        qv = vecs[0]
        return {
            "object": "list",
            "model": runner.served_name,
            "data": [{"object": "score", "index": i, "score": cos(qv, v)}
                     for i, v in enumerate(vecs[1:])],
            "usage": {"prompt_tokens": sum(len(p) for p in prompts),
                      "total_tokens": sum(len(p) for p in prompts)},
        }

    @app.post("/v1/messages")
    async def anthropic_messages(request: Request):
        """Anthropic-style Messages API (reference routes /v1/messages
        through its gateway: gateway/utils.py anthropic_model_exact +
        gateway/__init__.py supported_anthropic_routes)."""
        body = await request.json()
        messages = body.get("messages") or []
        system = body.get("system")
        norm = []
        if system:
            if isinstance(system, list):  # content-block form
                system = "".join(b.get("text", "") for b in system
                                 if isinstance(b, dict))
            norm.append({"role": "system", "content": system})
        for m in messages:
            c = m.get("content", "")
            if isinstance(c, list):
                c = "".join(b.get("text", "") for b in c
                            if isinstance(b, dict) and b.get("type") == "text")
            norm.append({"role": m.get("role", "user"), "content": c})
        tok = runner.tokenizer
        if hasattr(tok, "apply_chat_template"):
            try:
                prompt = tok.apply_chat_template(norm, tokenize=False,
                                                 add_generation_prompt=True)
            except TypeError:
                prompt = tok.apply_chat_template(norm)
        else:
            prompt = "\n".join(f"{m['role']}: {m['content']}" for m in norm)
        ids = tok.encode(prompt)
        if hasattr(ids, "ids"):
            ids = ids.ids
        ids = list(ids)
        oa = {
            "max_tokens": body.get("max_tokens", 1024),
            "temperature": body.get("temperature", 1.0),
            "top_p": body.get("top_p", 1.0),
            "stop": body.get("stop_sequences"),
        }
        params = _sampling_params(oa, runner.engine.cfg.spec.eos_token_id)
        _apply_lora_routing(params, body, runner)
        stop_strs = _stop_strings(oa)
        rid, q = runner.submit(ids, params)
        model_name = runner.served_name
        msg_id = f"msg_{rid}"

        def stop_reason(fin: str) -> str:
            return {"length": "max_tokens", "stop": "end_turn"}.get(fin, "end_turn")

        if body.get("stream"):
            async def gen():
                tokens: list[int] = []
                sent_len = 0
                fin = "end_turn"
                try:
                    start = {"type": "message_start", "message": {
                        "id": msg_id, "type": "message", "role": "assistant",
                        "model": model_name, "content": [],
                        "usage": {"input_tokens": len(ids), "output_tokens": 0}}}
                    yield f"event: message_start\ndata: {json.dumps(start)}\n\n"
                    cbs = {"type": "content_block_start", "index": 0,
                           "content_block": {"type": "text", "text": ""}}
                    yield f"event: content_block_start\ndata: {json.dumps(cbs)}\n\n"
                    while True:
                        if await request.is_disconnected():
                            runner.abort(rid)
                            return
                        item = await q.get()
                        if "error" in item:
                            err = {"type": "error",
                                   "error": {"type": "api_error",
                                             "message": item["error"]}}
                            yield f"event: error\ndata: {json.dumps(err)}\n\n"
                            return
                        tokens.append(item["token_id"])
                        text = runner.tokenizer.decode(tokens)
                        new = text[sent_len:]
                        if new and not new.endswith("\ufffd"):
                            sent_len = len(text)
                            d = {"type": "content_block_delta", "index": 0,
                                 "delta": {"type": "text_delta", "text": new}}
                            yield f"event: content_block_delta\ndata: {json.dumps(d)}\n\n"
                        if item["finished"]:
                            fin = stop_reason(item.get("finish_reason") or "stop")
                            break
                    yield ('event: content_block_stop\ndata: '
                           + json.dumps({"type": "content_block_stop", "index": 0})
                           + "\n\n")
                    md = {"type": "message_delta",
                          "delta": {"stop_reason": fin, "stop_sequence": None},
                          "usage": {"output_tokens": len(tokens)}}
                    yield f"event: message_delta\ndata: {json.dumps(md)}\n\n"
                    yield ('event: message_stop\ndata: '
                           + json.dumps({"type": "message_stop"}) + "\n\n")
                finally:
                    runner.release(rid)

            return StreamingResponse(gen(), media_type="text/event-stream")

        tokens = []
        fin = "stop"
        try:
            while True:
                item = await q.get()
                if "error" in item:
                    raise HTTPException(500, item["error"])
                tokens.append(item["token_id"])
                if stop_strs:
                    t = runner.tokenizer.decode(tokens)
                    if any(ss in t for ss in stop_strs):
                        runner.abort(rid)
                        fin = "stop_sequence"
                        break
                if item["finished"]:
                    fin = item.get("finish_reason") or "stop"
                    break
        finally:
            runner.release(rid)
        text = runner.tokenizer.decode(tokens)
        if stop_strs:
            cuts = [text.find(ss) for ss in stop_strs if ss in text]
            if cuts:
                text = text[:min(cuts)]
        sr = ("stop_sequence" if fin == "stop_sequence" else stop_reason(fin))
        return JSONResponse({
            "id": msg_id, "type": "message", "role": "assistant",
            "model": model_name,
            "content": [{"type": "text", "text": text}],
            "stop_reason": sr, "stop_sequence": None,
            "usage": {"input_tokens": len(ids), "output_tokens": len(tokens)},
        })

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        body = await request.json()
        inputs = body.get("input", [])
        if isinstance(inputs, str):
            inputs = [inputs]
        tok = runner.tokenizer
        prompts = []
        for item in inputs:
            if isinstance(item, str):
                ids = tok.encode(item)
                if hasattr(ids, "ids"):
                    ids = ids.ids
                prompts.append(list(ids) or [0])
            else:
                prompts.append([int(t) for t in item] or [0])
        pooling = body.get("pooling", "last")
        vecs = await runner.run_aux(runner.engine.runner.embed, prompts, pooling)
        return {
            "object": "list",
            "model": runner.served_name,
            "data": [{"object": "embedding", "index": i, "embedding": v}
                     for i, v in enumerate(vecs)],
            "usage": {"prompt_tokens": sum(len(p) for p in prompts),
                      "total_tokens": sum(len(p) for p in prompts)},
        }

    @app.post("/v1/rerank")
    @app.post("/rerank")
    async def rerank(request: Request):
        """Query-document relevance ranking (reference serves rerank models
        via vox-box/vLLM backends, routes/openai_compatible: /v1/rerank;
        first-party design: score = cosine similarity of last-token pooled
        embeddings from the same engine — no second model process)."""
        body = await request.json()
        query = body.get("query", "")
        docs = body.get("documents", [])
        top_n = int(body.get("top_n", len(docs)) or len(docs))
        tok = runner.tokenizer

        def enc(text):
            ids = tok.encode(text)
            if hasattr(ids, "ids"):
                ids = ids.ids
            return list(ids) or [0]

        if body.get("mode") == "generative" or "instruction" in body:
            # Qwen3-Reranker-style scoring: the model judges each
            # (instruction, query, doc) prompt and relevance is
            # P(yes | prompt) from the last-token logits
            import math as _math

            instr = body.get("instruction") or (
                "Judge whether the Document meets the requirements based "
                "on the Query. Answer only yes or no.")
            yes_id = enc(body.get("yes_token", "yes"))[-1]
            no_id = enc(body.get("no_token", "no"))[-1]
            prompts = [
                enc(f"{instr}\nQuery: {query}\nDocument: {d}\nAnswer:")
                for d in docs
            ]
            pairs = await runner.run_aux(
                runner.engine.runner.score_tokens, prompts, [yes_id, no_id])
            scored = [{"index": i,
                       "relevance_score":
                           1.0 / (1.0 + _math.exp(-(y - n))),
                       "document": {"text": docs[i]}}
                      for i, (y, n) in enumerate(pairs)]
            scored.sort(key=lambda r: -r["relevance_score"])
            ntok = sum(len(p) for p in prompts)
            return {"model": runner.served_name, "results": scored[:top_n],
                    "usage": {"prompt_tokens": ntok, "total_tokens": ntok}}

        prompts = [enc(query)] + [enc(d) for d in docs]
        vecs = await runner.run_aux(runner.engine.runner.embed, prompts, "mean")
        import math

        def cos(a, b):
            num = sum(x * y for x, y in zip(a, b))
            den = math.sqrt(sum(x * x for x in a)) * math.sqrt(sum(y * y for y in b))
            return num / den if den else 0.0

        qv = vecs[0]
        scored = [{"index": i, "relevance_score": cos(qv, v),
                   "document": {"text": docs[i]}}
                  for i, v in enumerate(vecs[1:])]
        scored.sort(key=lambda r: -r["relevance_score"])
        return {
            "model": runner.served_name,
            "results": scored[:top_n],
            "usage": {"prompt_tokens": sum(len(p) for p in prompts),
                      "total_tokens": sum(len(p) for p in prompts)},
        }

    @app.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = prompt[0] if prompt else ""
        tok = runner.tokenizer
        if isinstance(prompt, str):
            ids = tok.encode(prompt)
            if hasattr(ids, "ids"):
                ids = ids.ids
            ids = list(ids)
        else:
            ids = [int(t) for t in prompt]  # pre-tokenized
        echo = body.get("echo") and isinstance(prompt, str)
        return await _generate(request, body, ids, "text",
                               echo_text_prefix=prompt if echo else "")

    @app.post("/v1/responses")
    async def responses(request: Request):
        """OpenAI Responses API, minimal non-stream shape (reference
        endpoint registry includes /responses, gateway/utils.py:167-194):
        `input` is a string or message list; answer comes back as one
        output_text item."""
        body = await request.json()
        inp = body.get("input", "")
        messages = []
        if body.get("instructions"):
            messages.append({"role": "system",
                             "content": body["instructions"]})
        if isinstance(inp, str):
            messages.append({"role": "user", "content": inp})
        else:
            for m in inp or []:
                content = m.get("content")
                if isinstance(content, list):  # typed content parts
                    content = "".join(p.get("text", "") for p in content)
                messages.append({"role": m.get("role", "user"),
                                 "content": content or ""})
        tok = runner.tokenizer
        if hasattr(tok, "apply_chat_template"):
            try:
                prompt = tok.apply_chat_template(messages, tokenize=False,
                                                 add_generation_prompt=True)
            except TypeError:
                prompt = tok.apply_chat_template(messages)
        else:
            prompt = "\n".join(f"{m['role']}: {m['content']}" for m in messages)
        ids = tok.encode(prompt)
        if hasattr(ids, "ids"):
            ids = ids.ids
        inner = dict(body)
        inner["max_tokens"] = body.get("max_output_tokens") or body.get(
            "max_tokens") or 256
        inner.pop("stream", None)  # minimal surface: non-streaming
        resp = await _generate(request, inner, list(ids), "chat")
        data = json.loads(bytes(resp.body))
        msg = data["choices"][0]["message"]
        return JSONResponse({
            "id": data["id"].replace("req-", "resp_"),
            "object": "response",
            "created_at": data["created"],
            "model": data["model"],
            "status": "completed",
            "output": [{
                "type": "message", "role": "assistant",
                "content": [{"type": "output_text",
                             "text": msg.get("content") or ""}],
            }],
            "output_text": msg.get("content") or "",
            "usage": {
                "input_tokens": data["usage"]["prompt_tokens"],
                "output_tokens": data["usage"]["completion_tokens"],
                "total_tokens": data["usage"]["total_tokens"],
            },
        })

    @app.post("/v1/messages/count_tokens")
    async def count_tokens(request: Request):
        """Anthropic count_tokens (reference endpoint surface)."""
        body = await request.json()
        tok = runner.tokenizer
        text = "".join(
            (m.get("content") if isinstance(m.get("content"), str)
             else "".join(p.get("text", "") for p in m.get("content") or []))
            for m in body.get("messages") or [])
        if body.get("system"):
            text = str(body["system"]) + text
        ids = tok.encode(text)
        if hasattr(ids, "ids"):
            ids = ids.ids
        return {"input_tokens": len(list(ids))}

    return app


def _spawn_followers(argv_base: list[str], ranks: list[int], master_port: int):
    """The per-worker parent spawns follower processes for its local ranks
    (the first-party replacement for the reference's multi-process executor
    bootstrap — serve_manager.py:1685-1737 port bands + vLLM --headless
    followers / ranktables)."""
    import subprocess
    import sys

    procs = []
    for r in ranks:
        cmd = [sys.executable, "-m", "gpustack_amd.worker.engine_server",
               *argv_base, "--tp-rank", str(r), "--master-port", str(master_port)]
        procs.append(subprocess.Popen(cmd))
    return procs


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--served-name", required=True)
    ap.add_argument("--source", default="preset")
    ap.add_argument("--model-ref", required=True)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, required=True)
    ap.add_argument("--max-model-len", type=int, default=8192)
    ap.add_argument("--max-num-seqs", type=int, default=256)
    ap.add_argument("--gpu-memory-utilization", type=float, default=0.9)
    ap.add_argument("--device", default=None)
    ap.add_argument("--kv-cache-blocks", type=int, default=None)
    ap.add_argument("--backend-parameters", default="{}")
    ap.add_argument("--tp", type=int, default=1)
    ap.add_argument("--pp", type=int, default=1,
                    help="pipeline stages (layers partitioned across ranks; "
                         "global rank = pp_rank * tp + tp_rank)")
    ap.add_argument("--cp", type=int, default=1,
                    help="prefill context parallelism (prompt rows chunked "
                         "across ranks, per-layer KV all-gather, replicated "
                         "decode; vLLM --prefill-context-parallel-size "
                         "analog; mutually exclusive with --pp)")
    ap.add_argument("--tp-rank", type=int, default=None,
                    help="absolute rank of THIS process (set for spawned "
                         "followers; unset = per-worker parent)")
    ap.add_argument("--rank-base", type=int, default=0,
                    help="first rank hosted on this worker")
    ap.add_argument("--local-ranks", type=int, default=None,
                    help="ranks hosted on this worker (default: tp)")
    ap.add_argument("--master-addr", default="127.0.0.1")
    ap.add_argument("--master-port", type=int, default=None)
    args = ap.parse_args()

    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s %(name)s: %(message)s")
    import torch

    from ..engine import EngineConfig

    use_cuda = torch.cuda.is_available()
    device = args.device or ("cuda" if use_cuda else "cpu")
    comm = None
    followers = []
    world = args.tp * args.pp * args.cp
    if args.local_ranks is None:
        args.local_ranks = world if args.tp_rank is None else 1
    if world > 1:
        from ..parallel import init_parallel

        if args.tp_rank is None:
            # per-worker parent: hosts ranks [rank_base, rank_base+local)
            if args.master_port is None:
                import socket

                s = socket.socket()
                s.bind(("127.0.0.1", 0))
                args.master_port = s.getsockname()[1]
                s.close()
            argv_base = [
                "--served-name", args.served_name, "--source", args.source,
                "--model-ref", args.model_ref, "--port", str(args.port),
                "--max-model-len", str(args.max_model_len),
                "--max-num-seqs", str(args.max_num_seqs),
                "--gpu-memory-utilization", str(args.gpu_memory_utilization),
                "--backend-parameters", args.backend_parameters,
                "--tp", str(args.tp), "--pp", str(args.pp),
                "--cp", str(args.cp),
                "--rank-base", str(args.rank_base),
                "--master-addr", args.master_addr,
            ]
            if args.kv_cache_blocks:
                argv_base += ["--kv-cache-blocks", str(args.kv_cache_blocks)]
            if args.device:
                argv_base += ["--device", args.device]
            follow_ranks = list(range(args.rank_base + 1,
                                      args.rank_base + args.local_ranks))
            followers = _spawn_followers(argv_base, follow_ranks, args.master_port)
            args.tp_rank = args.rank_base
        local_ordinal = args.tp_rank - args.rank_base
        if use_cuda:
            device = f"cuda:{local_ordinal}"
            torch.cuda.set_device(local_ordinal)
        comm = init_parallel(args.tp, args.pp, args.tp_rank,
                             master_port=args.master_port,
                             device_id=local_ordinal if use_cuda else None,
                             master_addr=args.master_addr,
                             cp_size=args.cp)

    from pathlib import Path as _Path

    _cand = _Path(args.model_ref)
    if _cand.is_dir() and (_cand / "config.json").exists():
        import json as _json

        with open(_cand / "config.json") as _f:
            _arch = (_json.load(_f).get("architectures") or [""])[0]
        from ..models.encoder import is_encoder_arch

        if is_encoder_arch(_arch):
            app = create_encoder_app(str(_cand), args.served_name, device)
            import uvicorn

            uvicorn.run(app, host=args.host, port=args.port,
                        log_level="warning")
            return

    extra = json.loads(args.backend_parameters)
    cfg_kwargs = dict(
        model=args.model_ref,
        device=device,
        max_model_len=args.max_model_len,
        max_num_seqs=args.max_num_seqs,
        gpu_memory_utilization=args.gpu_memory_utilization,
        kv_cache_blocks=args.kv_cache_blocks,
        tp_size=args.tp,
        tp_rank=(args.tp_rank or 0) % args.tp,
    )
    if device == "cpu" and args.kv_cache_blocks is None:
        cfg_kwargs["kv_cache_blocks"] = 1024
    cfg_kwargs.update({k: v for k, v in extra.items() if k in EngineConfig.__dataclass_fields__})
    ecfg = EngineConfig(**cfg_kwargs)
    if args.source == "local_path":
        ecfg.enforce_random_weights = False
        if args.model_ref.endswith(".gguf"):
            ecfg.gguf_path = args.model_ref  # dequant-on-load execution
        else:
            ecfg.model_dir = args.model_ref

    if world > 1 and args.tp_rank != 0:
        # follower rank: no HTTP; run the coordinated engine loop forever
        import time as _time

        from ..engine import LLMEngine

        eng = LLMEngine(ecfg, comm)
        logger.info("TP follower rank %d ready", args.tp_rank)
        while True:
            if eng.tp_active():
                eng.step()
            else:
                _time.sleep(0.02)

    runner = EngineRunner(ecfg, args.served_name, comm)
    for ad in (extra.get("lora_adapters") or []):
        try:
            runner.engine.add_lora(ad["name"], ad["path"])
            logger.info("mounted LoRA adapter %s from %s", ad["name"], ad["path"])
        except Exception:  # noqa: BLE001
            logger.exception("failed to load LoRA adapter %s", ad.get("name"))
    app = create_app(runner)
    import uvicorn

    try:
        uvicorn.run(app, host=args.host, port=args.port, log_level="warning")
    finally:
        for p in followers:
            p.terminate()


if __name__ == "__main__":
    main()
