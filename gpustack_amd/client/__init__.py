"""Typed HTTP client for server<->worker communication
(reference: gpustack/client/generated_clientset.py + watch mechanism)."""
from __future__ import annotations

import json
import logging
from typing import Iterator

import httpx

logger = logging.getLogger(__name__)


class ServerClient:
    def __init__(self, base_url: str, token: str | None = None, timeout: float = 30.0):
        self.base_url = base_url.rstrip("/")
        headers = {}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._c = httpx.Client(base_url=self.base_url, headers=headers, timeout=timeout)

    # -- worker lifecycle --
    def register_worker(self, payload: dict) -> dict:
        r = self._c.post("/v2/workers/register", json=payload)
        r.raise_for_status()
        return r.json()

    def worker_heartbeat(self, worker_id: int) -> None:
        self._c.post(f"/v2/workers/{worker_id}/heartbeat").raise_for_status()

    def worker_status(self, worker_id: int, status: dict) -> None:
        self._c.post(f"/v2/workers/{worker_id}/status", json={"status": status}).raise_for_status()

    # -- instances --
    def list_instances(self, worker_id: int | None = None) -> list[dict]:
        params = {}
        if worker_id is not None:
            params["worker_id"] = worker_id
        r = self._c.get("/v2/model_instances", params=params)
        r.raise_for_status()
        return r.json()["items"]

    def update_instance(self, instance_id: int, **fields) -> dict:
        r = self._c.patch(f"/v2/model_instances/{instance_id}", json=fields)
        r.raise_for_status()
        return r.json()

    def get_model(self, model_id: int) -> dict:
        r = self._c.get(f"/v2/models/{model_id}")
        r.raise_for_status()
        return r.json()

    def watch_instances(self, worker_id: int) -> Iterator[dict]:
        """Long-lived NDJSON watch stream (server replays snapshot then
        relays bus events; reference: client watch at
        generated_model_instance_client.py:165)."""
        with self._c.stream(
            "GET", "/v2/model_instances",
            params={"watch": "true", "worker_id": worker_id},
            timeout=httpx.Timeout(30.0, read=None),
        ) as resp:
            resp.raise_for_status()
            for line in resp.iter_lines():
                if not line:
                    continue
                try:
                    yield json.loads(line)
                except json.JSONDecodeError:
                    logger.warning("bad watch frame: %r", line[:200])

    def close(self):
        self._c.close()
