"""OpenAI-compatible inference API + proxy (reference: gpustack/routes/openai.py:185).

Resolves the requested model through model routes (weighted targets) or
directly by model name, picks a RUNNING instance round-robin
(reference LB: gpustack/http_proxy/load_balancer.py:7), relays the request
to the worker-side engine server, streams SSE chunks back, and records
token usage (reference: api/middlewares.py:81 ModelUsageMiddleware).
"""
from __future__ import annotations

import itertools
import json
import logging
import random
import time

import httpx
from fastapi import APIRouter, Depends, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from ..db import get_session
from ..schemas import (
    Model, ModelInstance, ModelInstanceState, ModelProvider, ModelRoute,
    ModelUsage, User, Worker,
)
from .deps import get_current_user

logger = logging.getLogger(__name__)

router = APIRouter()

_rr = itertools.count()

PROXY_TIMEOUT = 1800.0

# one registry shared with the gateway (reference: gateway/utils.py:167-194)
OPENAI_PATHS = [
    "/v1/chat/completions",
    "/v1/completions",
    "/v1/responses",
    "/v1/embeddings",
    "/v1/rerank",
    "/v1/score",
    "/v1/messages",
    "/v1/messages/count_tokens",
]


def _resolve_model_name(name: str) -> tuple[str, str | None]:
    """Model-route indirection with weighted targets, plus per-LoRA child
    routes (reference: gpustack/server/lora_model_routes.py): requesting an
    adapter name from some model's `lora_adapters` routes to that model's
    instances with the adapter applied per-request. Returns
    (target model name, lora adapter name or None)."""
    with get_session() as s:
        route = s.query(ModelRoute).filter_by(name=name).first()
        if route and route.targets:
            weights = [t.get("weight", 1) for t in route.targets]
            pick = random.choices(route.targets, weights=weights)[0]
            return pick.get("model_name", name), None
        if s.query(Model).filter_by(name=name).first() is None:
            for m in s.query(Model).filter(Model.lora_adapters.isnot(None)).all():
                for ad in m.lora_adapters or []:
                    if ad.get("name") == name:
                        return m.name, name
    return name, None


def _find_provider(model_name: str):
    """External provider fallback (reference: schemas/model_provider.py —
    providers routable through the same gateway)."""
    with get_session() as s:
        for p in s.query(ModelProvider).filter_by(enabled=True).all():
            if p.models is None or model_name in p.models:
                return p.to_dict()
    return None


def _pick_instance(model_name: str, user: User | None = None) -> tuple[Model, dict]:
    from .deps import model_allowed_for_user

    with get_session() as s:
        model = s.query(Model).filter_by(name=model_name).first()
        if not model:
            raise HTTPException(404, f"model {model_name!r} not found")
        if user is not None and not model_allowed_for_user(user, model):
            # 404 (not 403): tenants must not learn other orgs' model names
            raise HTTPException(404, f"model {model_name!r} not found")
        insts = (
            s.query(ModelInstance)
            .filter_by(model_id=model.id, state=ModelInstanceState.RUNNING.value)
            .all()
        )
        if not insts:
            raise HTTPException(
                503, f"no running instances for model {model_name!r}"
            )
        pick = insts[next(_rr) % len(insts)]  # round-robin
        return model, pick.to_dict()


def _record_usage(user: User, model: Model, usage: dict | None) -> None:
    if not usage:
        return
    try:
        with get_session() as s:
            date = time.strftime("%Y-%m-%d")
            row = (
                s.query(ModelUsage)
                .filter_by(user_id=user.id, model_name=model.name, date=date)
                .first()
            )
            if row is None:
                row = ModelUsage(user_id=user.id, model_id=model.id,
                                 model_name=model.name, date=date,
                                 prompt_tokens=0, completion_tokens=0,
                                 request_count=0)
                s.add(row)
            row.prompt_tokens += usage.get("prompt_tokens", 0)
            row.completion_tokens += usage.get("completion_tokens", 0)
            row.request_count += 1
            s.commit()
    except Exception:  # noqa: BLE001
        logger.exception("usage recording failed")


async def _proxy(request: Request, path: str, user: User):
    body_bytes = await request.body()
    try:
        body = json.loads(body_bytes or b"{}")
    except json.JSONDecodeError:
        raise HTTPException(400, "invalid JSON body")
    name = body.get("model")
    if not name:
        raise HTTPException(400, "missing 'model'")
    target_name, lora_name = _resolve_model_name(name)
    headers = {}
    with get_session() as s:
        local = s.query(Model).filter_by(name=target_name).first() is not None
    tunnel_worker = None
    if local:
        model, inst = _pick_instance(target_name, user)
        with get_session() as s:
            w = s.get(Worker, inst["worker_id"]) if inst.get("worker_id") else None
            if w is not None and w.proxy_mode == "tunnel":
                tunnel_worker = (w.id, inst["port"])
        url = f"http://{inst['worker_ip']}:{inst['port']}{path}"
    else:
        provider = _find_provider(target_name)
        if provider is None:
            raise HTTPException(404, f"model {target_name!r} not found")
        model = Model(id=0, name=f"{provider['name']}/{target_name}")
        url = provider["base_url"].rstrip("/") + path.replace("/v1", "", 1) \
            if provider["base_url"].rstrip("/").endswith("/v1") \
            else provider["base_url"].rstrip("/") + path
        if provider.get("api_key"):
            headers["Authorization"] = f"Bearer {provider['api_key']}"
    body["model"] = target_name
    if lora_name:
        body["lora_name"] = lora_name
    stream = bool(body.get("stream"))
    if tunnel_worker is not None:
        return await _proxy_via_tunnel(tunnel_worker, path, body, user, model, stream)
    client = httpx.AsyncClient(timeout=PROXY_TIMEOUT)
    req = client.build_request("POST", url, json=body, headers=headers)
    try:
        resp = await client.send(req, stream=stream)
    except httpx.ConnectError:
        await client.aclose()
        raise HTTPException(502, "instance unreachable")
    if not stream:
        data = await resp.aread()
        await client.aclose()
        try:
            payload = json.loads(data)
            _record_usage(user, model, payload.get("usage"))
        except Exception:  # noqa: BLE001
            payload = None
        return JSONResponse(
            content=payload if payload is not None else {"raw": data.decode(errors="replace")},
            status_code=resp.status_code,
        )

    async def relay():
        usage = None
        try:
            async for chunk in resp.aiter_lines():
                if chunk.startswith("data:"):
                    frag = chunk[5:].strip()
                    if frag and frag != "[DONE]":
                        try:
                            u = json.loads(frag).get("usage")
                            if u:
                                usage = u
                        except json.JSONDecodeError:
                            pass
                yield (chunk + "\n\n").encode()
        except httpx.HTTPError as e:  # mid-stream error frame (openai.py:423-483)
            yield f"data: {json.dumps({'error': {'message': str(e)}})}\n\n".encode()
        finally:
            await resp.aclose()
            await client.aclose()
            _record_usage(user, model, usage)

    return StreamingResponse(relay(), media_type="text/event-stream",
                             status_code=resp.status_code)


async def _proxy_via_tunnel(tw, path: str, body: dict, user: User,
                            model: Model, stream: bool):
    """Relay through the worker-initiated long-poll tunnel
    (reference: websocket_proxy/proxy_server.py semantics)."""
    from .tunnel import hub

    worker_id, port = tw
    payload = json.dumps(body).encode()
    try:
        status, ctype, chunks = await hub.request(
            worker_id, port, "POST", path, payload,
            {"content-type": "application/json"},
        )
    except TimeoutError:
        raise HTTPException(502, "tunnel worker did not answer")
    if not stream:
        data = b"".join([c async for c in chunks])
        try:
            parsed = json.loads(data)
            _record_usage(user, model, parsed.get("usage"))
        except Exception:  # noqa: BLE001
            parsed = {"raw": data.decode(errors="replace")}
        return JSONResponse(parsed, status_code=status)

    async def relay():
        buf = b""
        usage = None
        async for c in chunks:
            buf += c
            yield c
        for line in buf.decode(errors="replace").splitlines():
            if line.startswith("data:"):
                frag = line[5:].strip()
                if frag and frag != "[DONE]":
                    try:
                        u = json.loads(frag).get("usage")
                        if u:
                            usage = u
                    except json.JSONDecodeError:
                        pass
        _record_usage(user, model, usage)

    return StreamingResponse(relay(), media_type=ctype, status_code=status)


@router.get("/v1/models")
def list_models_v1(user: User = Depends(get_current_user)):
    from .deps import model_allowed_for_user

    with get_session() as s:
        models = [m for m in s.query(Model).all()
                  if model_allowed_for_user(user, m)]
        routes = s.query(ModelRoute).all()
        items = [
            {"id": m.name, "object": "model", "created": int(m.created_at),
             "owned_by": "gpustack_amd"}
            for m in models
        ] + [
            {"id": ad["name"], "object": "model", "created": int(m.created_at),
             "owned_by": "gpustack_amd", "parent": m.name}
            for m in models for ad in (m.lora_adapters or [])
            if ad.get("name")
        ] + [
            {"id": r.name, "object": "model", "created": int(r.created_at),
             "owned_by": "gpustack_amd/route"}
            for r in routes
        ] + [
            {"id": m, "object": "model", "created": int(p.created_at),
             "owned_by": f"provider/{p.name}"}
            for p in s.query(ModelProvider).filter_by(enabled=True).all()
            for m in (p.models or [])
        ]
    return {"object": "list", "data": items}


@router.post("/v1/chat/completions")
async def chat_completions(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/chat/completions", user)


@router.post("/v1/completions")
async def completions(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/completions", user)


@router.post("/v1/embeddings")
async def embeddings(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/embeddings", user)


@router.post("/v1/rerank")
async def rerank(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/rerank", user)


# ---- descoped modality surface -------------------------------------------
# The reference proxies audio (STT/TTS via vox-box), image generation and
# moderations to external engines (gateway/utils.py:167-194,
# worker/backends/vox_box.py). This framework's first-party engine serves
# text model families only; the endpoints exist wire-compatibly and return
# a structured 501 naming the gap (formal descope — PARITY.md "audio/image
# modalities") instead of a bare 404, so OpenAI SDK clients fail cleanly.
_UNSUPPORTED_MODALITIES = [
    "/v1/audio/transcriptions", "/v1/audio/translations", "/v1/audio/speech",
    "/v1/images/generations", "/v1/images/edits", "/v1/images/variations",
    "/v1/moderations",
]


def _register_unsupported(path: str) -> None:
    @router.post(path, name=f"unsupported{path.replace('/', '_')}")
    async def _unsupported(user: User = Depends(get_current_user)):
        raise HTTPException(
            501,
            detail={
                "error": {
                    "message": f"{path} is not supported: this deployment "
                               "serves text model families (llm, embedding, "
                               "reranker) on the first-party MI355X engine; "
                               "audio/image modalities are descoped",
                    "type": "unsupported_modality",
                }
            },
        )


for _p in _UNSUPPORTED_MODALITIES:
    _register_unsupported(_p)


@router.post("/v1/messages")
async def anthropic_messages(request: Request, user: User = Depends(get_current_user)):
    """Anthropic-style Messages API, proxied to the placed instance
    (reference gateway routes /v1/messages: gateway/__init__.py:70-75)."""
    return await _proxy(request, "/v1/messages", user)


@router.post("/v1/score")
async def score(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/score", user)


@router.post("/v1/responses")
async def responses(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/responses", user)


# frozen legacy mount (reference: routes/openai.py:81-92 re-mounts a legacy
# subset at /v1-openai for older SDK base_url conventions)
@router.post("/v1-openai/chat/completions")
async def legacy_chat(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/chat/completions", user)


@router.post("/v1-openai/completions")
async def legacy_completions(request: Request,
                             user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/completions", user)


@router.post("/v1-openai/embeddings")
async def legacy_embeddings(request: Request,
                            user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/embeddings", user)


@router.get("/v1-openai/models")
def legacy_models(user: User = Depends(get_current_user)):
    return list_models_v1(user)


@router.post("/v1/messages/count_tokens")
async def count_tokens(request: Request, user: User = Depends(get_current_user)):
    return await _proxy(request, "/v1/messages/count_tokens", user)
