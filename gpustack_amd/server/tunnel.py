"""Tunnel proxy for workers behind NAT (reference: gpustack/websocket_proxy/
+ http_proxy — worker-initiated connection carries inference traffic).

Re-designed as an HTTP long-poll tunnel (no extra dependencies, plain
httpx on the worker side):

  worker loop:   GET  /v2/tunnel/jobs?worker_id=N   (long-poll, worker token)
                 -> job {id, port, method, path, headers, body_b64} | 204
  worker reply:  POST /v2/tunnel/reply/{id}  (streamed raw body,
                 X-Tunnel-Status / X-Tunnel-Content-Type headers)

The server-side gateway calls TunnelHub.request(); response bytes stream
through the reply upload straight into the waiting client response, so SSE
token streams relay with no buffering.
"""
from __future__ import annotations

import asyncio
import base64
import logging
import time
import uuid

logger = logging.getLogger(__name__)

JOB_POLL_TIMEOUT = 20.0
REPLY_TIMEOUT = 120.0


class TunnelHub:
    def __init__(self):
        self._jobs: dict[int, asyncio.Queue] = {}
        self._replies: dict[str, asyncio.Queue] = {}
        self._last_poll: dict[int, float] = {}

    def worker_connected(self, worker_id: int) -> bool:
        return time.time() - self._last_poll.get(worker_id, 0) < JOB_POLL_TIMEOUT * 2

    def _job_q(self, worker_id: int) -> asyncio.Queue:
        if worker_id not in self._jobs:
            self._jobs[worker_id] = asyncio.Queue()
        return self._jobs[worker_id]

    async def next_job(self, worker_id: int) -> dict | None:
        self._last_poll[worker_id] = time.time()
        try:
            return await asyncio.wait_for(self._job_q(worker_id).get(),
                                          timeout=JOB_POLL_TIMEOUT)
        except asyncio.TimeoutError:
            return None

    async def next_jobs(self, worker_id: int, limit: int = 32) -> list[dict]:
        """Batch pickup: block for the first job, then drain whatever else
        is queued — dispatch rate is no longer one job per poll round-trip
        (the r1 concurrent-streaming-load concern)."""
        first = await self.next_job(worker_id)
        if first is None:
            return []
        out = [first]
        q = self._job_q(worker_id)
        while len(out) < limit:
            try:
                out.append(q.get_nowait())
            except asyncio.QueueEmpty:
                break
        return out

    async def begin_reply(self, req_id: str, status: int, content_type: str):
        q = self._replies.get(req_id)
        if q is None:
            return None
        await q.put(("begin", status, content_type))
        return q

    async def request(self, worker_id: int, port: int, method: str, path: str,
                      body: bytes, headers: dict | None = None):
        """Send a request through the tunnel; returns (status, content_type,
        async-iterator of body chunks)."""
        req_id = uuid.uuid4().hex
        q: asyncio.Queue = asyncio.Queue()
        self._replies[req_id] = q
        await self._job_q(worker_id).put({
            "id": req_id, "port": port, "method": method, "path": path,
            "headers": headers or {},
            "body_b64": base64.b64encode(body).decode(),
        })
        try:
            kind, status, ctype = await asyncio.wait_for(q.get(), REPLY_TIMEOUT)
        except asyncio.TimeoutError:
            self._replies.pop(req_id, None)
            raise TimeoutError("tunnel reply timeout")
        assert kind == "begin"

        async def chunks():
            try:
                while True:
                    item = await asyncio.wait_for(q.get(), REPLY_TIMEOUT)
                    if item[0] == "chunk":
                        yield item[1]
                    else:  # end
                        return
            finally:
                self._replies.pop(req_id, None)

        return status, ctype, chunks()

    async def push_chunk(self, req_id: str, data: bytes) -> None:
        q = self._replies.get(req_id)
        if q is not None:
            await q.put(("chunk", data))

    async def end_reply(self, req_id: str) -> None:
        q = self._replies.get(req_id)
        if q is not None:
            await q.put(("end",))


hub = TunnelHub()
