"""Worker agent (reference: gpustack/worker/worker.py:65).

Registers with the server (retrying), runs heartbeat + status-sync
threads, the ServeManager watch/health loops, and a small worker API
(health, per-instance logs, Prometheus node/GPU metrics — reference
worker/exporter.py:43)."""
from __future__ import annotations

import logging
import socket
import threading
import time
from pathlib import Path

from fastapi import FastAPI, HTTPException
from fastapi.responses import PlainTextResponse, Response

from ..client import ServerClient
from ..config import Config
from .detector import collect_system_status
from .serve_manager import ServeManager

logger = logging.getLogger(__name__)


def _default_ip(server_url: str) -> str:
    try:
        host = server_url.split("//")[-1].split("/")[0].split(":")[0]
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.connect((host, 80))
        ip = s.getsockname()[0]
        s.close()
        return ip
    except OSError:
        return "127.0.0.1"


class WorkerAgent:
    def __init__(self, cfg: Config):
        assert cfg.server_url, "worker needs --server-url"
        self.cfg = cfg
        self.client = ServerClient(cfg.server_url, token=cfg.token)
        self.worker_id: int | None = None
        self.serve_manager: ServeManager | None = None
        self._stop = False

    def register(self) -> None:
        cfg = self.cfg
        name = cfg.worker_name or socket.gethostname()
        ip = cfg.worker_ip or _default_ip(cfg.server_url)
        status = collect_system_status(cfg.gpu_devices)
        delay = 2.0
        while not self._stop:
            try:
                w = self.client.register_worker({
                    "name": name,
                    "hostname": socket.gethostname(),
                    "ip": ip,
                    "port": cfg.worker_port,
                    "metrics_port": cfg.worker_metrics_port,
                    "labels": cfg.labels,
                    "status": status,
                    "system_reserved": cfg.system_reserved,
                    "proxy_mode": cfg.proxy_mode,
                })
                self.worker_id = w["id"]
                logger.info("registered as worker id=%s name=%s ip=%s", w["id"], name, ip)
                return
            except Exception as e:  # noqa: BLE001
                logger.warning("registration failed (%s); retrying in %.0fs", e, delay)
                time.sleep(delay)
                delay = min(delay * 2, 30)

    def heartbeat_loop(self) -> None:
        while not self._stop:
            try:
                self.client.worker_heartbeat(self.worker_id)
            except Exception as e:  # noqa: BLE001
                logger.warning("heartbeat failed: %s", e)
                if "404" in str(e):
                    self.register()
            time.sleep(self.cfg.heartbeat_interval)

    def status_loop(self) -> None:
        while not self._stop:
            try:
                status = collect_system_status(self.cfg.gpu_devices)
                self.client.worker_status(self.worker_id, status)
            except Exception as e:  # noqa: BLE001
                logger.warning("status sync failed: %s", e)
            time.sleep(self.cfg.worker_status_interval)

    def tunnel_loop(self) -> None:
        """NAT mode: long-poll the server for tunneled requests and stream
        local engine responses back (reference: websocket_proxy/message_client)."""
        import base64

        import httpx as _h

        headers = {"Authorization": f"Bearer {self.cfg.token}"}
        while not self._stop:
            try:
                r = self.client._c.get("/v2/tunnel/jobs",
                                       params={"worker_id": self.worker_id,
                                               "batch": 1},
                                       timeout=_h.Timeout(30.0, read=40.0))
                if r.status_code != 200:
                    continue
                for job in r.json().get("jobs", []):
                    threading.Thread(target=self._serve_tunnel_job,
                                     args=(job,), daemon=True).start()
            except Exception as e:  # noqa: BLE001
                logger.debug("tunnel poll error: %s", e)
                time.sleep(1)

    def _serve_tunnel_job(self, job: dict) -> None:
        import base64

        import httpx as _h

        body = base64.b64decode(job.get("body_b64", ""))
        url = f"http://127.0.0.1:{job['port']}{job['path']}"
        try:
            with _h.stream(job.get("method", "POST"), url, content=body,
                           headers=job.get("headers") or {},
                           timeout=_h.Timeout(30.0, read=None)) as resp:
                def gen():
                    for chunk in resp.iter_bytes():
                        yield chunk

                self.client._c.post(
                    f"/v2/tunnel/reply/{job['id']}",
                    content=gen(),
                    headers={
                        "X-Tunnel-Status": str(resp.status_code),
                        "X-Tunnel-Content-Type": resp.headers.get(
                            "content-type", "application/json"),
                    },
                    timeout=_h.Timeout(30.0, read=120.0),
                )
        except Exception as e:  # noqa: BLE001
            logger.warning("tunnel job %s failed: %s", job.get("id"), e)
            try:
                self.client._c.post(
                    f"/v2/tunnel/reply/{job['id']}",
                    content=b'{"error": {"message": "tunnel upstream failed"}}',
                    headers={"X-Tunnel-Status": "502",
                             "X-Tunnel-Content-Type": "application/json"},
                )
            except Exception:  # noqa: BLE001
                pass

    def create_api(self) -> FastAPI:
        app = FastAPI(title="gpustack_amd-worker")

        @app.get("/healthz")
        def healthz():
            return {"status": "ok", "worker_id": self.worker_id}

        @app.get("/files/model-config")
        def model_config(path: str):
            """Read a local model dir's config.json for the server/scheduler
            (reference: routes/worker/filesystem.py /files/model-config)."""
            import json as _json

            p = Path(path) / "config.json"
            if not p.exists():
                raise HTTPException(404, "config.json not found")
            return _json.loads(p.read_text())

        @app.get("/files/file-exists")
        def file_exists(path: str):
            p = Path(path)
            return {"exists": p.exists(), "is_dir": p.is_dir(),
                    "size": p.stat().st_size if p.is_file() else None}

        @app.get("/files/model-weight-size")
        def model_weight_size(path: str):
            total = 0
            p = Path(path)
            pats = ("*.safetensors",) if p.is_dir() else ()
            if p.is_file():
                total = p.stat().st_size
            else:
                for pat in pats + ("*.gguf",):
                    for f in p.glob(pat):
                        total += f.stat().st_size
            return {"weight_size_bytes": total}

        @app.get("/files/parse-gguf")
        def parse_gguf(path: str):
            """First-party in-process GGUF header parse (reference runs the
            gguf-parser Go binary remotely via the worker filesystem API,
            routes/worker/filesystem.py:105-309 + scheduler/calculator.py:
            622-682 — we parse natively, utils/gguf.py)."""
            from ..utils.gguf import read_gguf

            p = Path(path)
            if not p.is_file():
                raise HTTPException(404, "gguf file not found")
            try:
                info = read_gguf(p)
            except ValueError as e:
                raise HTTPException(422, str(e))
            meta = {k: v for k, v in info.metadata.items()
                    if not isinstance(v, list)}
            return {"version": info.version,
                    "architecture": info.architecture,
                    "n_params": info.n_params,
                    "weight_bytes": info.weight_bytes,
                    "n_tensors": len(info.tensors),
                    "metadata": meta}

        @app.get("/logs/{instance_name}")
        def logs(instance_name: str, tail: int = 200, follow: bool = False,
                 timeout_s: float = 300.0):
            """Instance log tail; `follow=true` streams appended lines as
            they land (reference: log_sources follow streaming) until
            timeout_s or the file disappears."""
            path = Path(self.cfg.data_dir) / "log" / "instances" / f"{instance_name}.log"
            if not path.exists():
                raise HTTPException(404, "no log for instance")
            if not follow:
                lines = path.read_text(errors="replace").splitlines()
                return PlainTextResponse("\n".join(lines[-tail:]))

            def stream():
                import time as _t

                with open(path, "rb") as f:
                    # serve the tail first, then poll for appended data
                    lines = f.read().decode(errors="replace").splitlines(True)
                    yield "".join(lines[-tail:])
                    deadline = _t.time() + timeout_s
                    while _t.time() < deadline:
                        chunk = f.read()
                        if chunk:
                            yield chunk.decode(errors="replace")
                        elif not path.exists():
                            return
                        else:
                            _t.sleep(0.5)

            from fastapi.responses import StreamingResponse

            return StreamingResponse(stream(), media_type="text/plain")

        @app.get("/metrics")
        def metrics():
            from prometheus_client import CollectorRegistry, Gauge, generate_latest

            reg = CollectorRegistry()
            st = collect_system_status(self.cfg.gpu_devices)
            g_cpu = Gauge("gpustack_worker_cpu_utilization_rate", "cpu util", registry=reg)
            g_cpu.set(st.get("cpu", {}).get("utilization_rate", 0))
            g_mem = Gauge("gpustack_worker_memory_used_bytes", "ram used", registry=reg)
            g_mem.set(st.get("memory", {}).get("used", 0))
            g_util = Gauge("gpustack_worker_gpu_utilization_rate", "gpu util",
                           ["index"], registry=reg)
            g_vt = Gauge("gpustack_worker_gpu_vram_total_bytes", "vram total",
                         ["index"], registry=reg)
            g_vu = Gauge("gpustack_worker_gpu_vram_used_bytes", "vram used",
                         ["index"], registry=reg)
            g_temp = Gauge("gpustack_worker_gpu_temperature_celsius", "temp",
                           ["index"], registry=reg)
            for d in st.get("gpu_devices", []):
                idx = str(d["index"])
                g_util.labels(idx).set(d.get("core", {}).get("utilization_rate", 0))
                g_vt.labels(idx).set(d.get("memory", {}).get("total", 0))
                g_vu.labels(idx).set(d.get("memory", {}).get("used", 0))
                g_temp.labels(idx).set(d.get("temperature", 0))
            body = generate_latest(reg).decode()
            # re-export each local engine's runtime metrics with an
            # instance label (reference: worker metrics aggregation of
            # backend runtime metrics)
            import httpx as _hx

            if self.serve_manager is not None:
                for ip in list(self.serve_manager.processes.values()):
                    if not ip.healthy:
                        continue
                    try:
                        r = _hx.get(f"http://127.0.0.1:{ip.port}/metrics",
                                    timeout=2)
                    except _hx.HTTPError:
                        continue
                    inst = ip.instance.get("name", "")
                    for ln in r.text.splitlines():
                        if ln.startswith("#"):
                            body += ln + "\n"
                        elif ln.strip():
                            name, _, val = ln.partition(" ")
                            body += f'{name}{{instance="{inst}"}} {val}\n'
            return Response(body, media_type="text/plain; version=0.0.4")

        return app

    def start(self) -> None:
        self.register()
        self.serve_manager = ServeManager(self.cfg, self.client, self.worker_id)
        from .benchmark_manager import BenchmarkManager

        self.benchmark_manager = BenchmarkManager(
            self.cfg, self.client, self.worker_id, self.serve_manager
        )
        threads = [
            threading.Thread(target=self.heartbeat_loop, name="heartbeat", daemon=True),
            threading.Thread(target=self.status_loop, name="status", daemon=True),
            threading.Thread(target=self.serve_manager.watch_loop, name="serve-watch", daemon=True),
            threading.Thread(target=self.serve_manager.health_loop, name="serve-health", daemon=True),
            threading.Thread(target=self.benchmark_manager.poll_loop, name="benchmarks", daemon=True),
        ]
        if self.cfg.proxy_mode == "tunnel":
            threads.append(threading.Thread(target=self.tunnel_loop,
                                            name="tunnel", daemon=True))
        for t in threads:
            t.start()
        import uvicorn

        app = self.create_api()
        logger.info("worker API on :%d", self.cfg.worker_port)
        uvicorn.run(app, host="0.0.0.0", port=self.cfg.worker_port, log_level="warning")

    def stop(self) -> None:
        self._stop = True
        if getattr(self, "benchmark_manager", None):
            self.benchmark_manager.stop()
        if self.serve_manager:
            self.serve_manager.stop()


def run_worker(cfg: Config) -> None:
    cfg.ensure_dirs()
    WorkerAgent(cfg).start()
