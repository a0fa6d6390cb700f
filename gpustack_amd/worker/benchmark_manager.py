"""Worker-side benchmark execution (reference: gpustack/worker/benchmark_manager.py:89).

Watches Benchmark rows assigned to this worker, drives the first-party
load generator (gpustack_amd.bench.loadgen) against the local RUNNING
instance of the target model, and posts the TTFT/TPOT/TPS aggregates back."""
from __future__ import annotations

import logging
import threading
import time

import httpx

from ..bench.analysis import analyze_profile
from ..bench.loadgen import LoadSpec, run_load
from ..client import ServerClient
from ..config import Config

logger = logging.getLogger(__name__)


class BenchmarkManager:
    def __init__(self, cfg: Config, client: ServerClient, worker_id: int,
                 serve_manager):
        self.cfg = cfg
        self.client = client
        self.worker_id = worker_id
        self.serve_manager = serve_manager
        self._stop = False
        self._running: set[int] = set()

    def stop(self):
        self._stop = True

    def poll_loop(self) -> None:
        while not self._stop:
            try:
                r = self.client._c.get("/v2/benchmarks")
                r.raise_for_status()
                for b in r.json()["items"]:
                    if (b.get("worker_id") == self.worker_id
                            and b.get("state") == "pending"
                            and b["id"] not in self._running):
                        self._running.add(b["id"])
                        threading.Thread(target=self._run_one, args=(b,),
                                         daemon=True).start()
            except httpx.HTTPError as e:
                logger.warning("benchmark poll failed: %s", e)
            time.sleep(3.0)

    def _find_instance_port(self, model_name: str) -> int | None:
        for ip in self.serve_manager.processes.values():
            if ip.instance.get("model_name") == model_name and ip.healthy:
                return ip.port
        # fall back: ask the server
        try:
            insts = self.client.list_instances(worker_id=self.worker_id)
            for i in insts:
                if i.get("model_name") == model_name and i.get("state") == "running":
                    return i.get("port")
        except httpx.HTTPError:
            pass
        return None

    def _run_one(self, b: dict) -> None:
        bid = b["id"]
        try:
            port = self._find_instance_port(b["model_name"])
            if port is None:
                raise RuntimeError("no local running instance for model")
            self._update(bid, state="running")
            cfgd = b.get("config") or {}
            # sweep profiles (reference: multi-point performance profiles):
            # config.sweep = [v1, v2, ...] runs one load point per value and
            # stores the whole curve under results.profile
            values = cfgd.get("sweep") or [cfgd.get("value", 8)]
            profile = []
            for v in values:
                spec = LoadSpec(
                    mode=cfgd.get("mode", "concurrency"),
                    value=float(v),
                    duration_s=float(cfgd.get("duration_s", 30)),
                    isl=int(cfgd.get("isl", 128)),
                    osl=int(cfgd.get("osl", 64)),
                    model=b["model_name"],
                )
                snap_before = self._snapshot(b["model_name"])
                point = run_load(f"http://127.0.0.1:{port}", spec,
                                 with_artifacts=bool(cfgd.get("artifacts")))
                point["value"] = float(v)
                # per-point worker/GPU/instance snapshot (reference:
                # schemas/benchmark.py:228-293 snapshot tables)
                point["snapshot"] = {"before": snap_before,
                                     "after": self._snapshot(b["model_name"])}
                profile.append(point)
            results = dict(max(profile, key=lambda r: r.get("output_tps") or 0))
            if len(profile) > 1:
                results["profile"] = profile
            # SLA + saturation analysis (reference: benchmark SLA thresholds
            # and saturation probe)
            results["analysis"] = analyze_profile(profile, cfgd.get("sla"))
            self._update(bid, state="completed", results=results)
            logger.info("benchmark %s done: %s out-tok/s, p50 TTFT %s ms",
                        b.get("name"), results.get("output_tps"),
                        results.get("ttft_p50_ms"))
        except Exception as e:  # noqa: BLE001
            logger.exception("benchmark %s failed", bid)
            self._update(bid, state="error", state_message=str(e))
        finally:
            self._running.discard(bid)

    def _snapshot(self, model_name: str) -> dict:
        """GPU + instance state at a load point (utilization, VRAM,
        instance pid/port) — the reference snapshots workers/GPUs/instances
        around every benchmark run."""
        out: dict = {"ts": time.time()}
        try:
            from .detector import detect_gpus

            gpus = detect_gpus(self.cfg.gpu_devices or None)
            out["gpus"] = [
                {"index": g.get("index"),
                 "utilization": (g.get("core") or {}).get("utilization_rate"),
                 "vram_used": (g.get("memory") or {}).get("used"),
                 "vram_total": (g.get("memory") or {}).get("total")}
                for g in gpus
            ]
        except Exception:  # noqa: BLE001
            out["gpus"] = []
        for ip in self.serve_manager.processes.values():
            if ip.instance.get("model_name") == model_name:
                out["instance"] = {"pid": ip.proc.pid, "port": ip.port,
                                   "healthy": ip.healthy}
                break
        return out

    def _update(self, bid: int, **fields) -> None:
        try:
            self.client._c.patch(f"/v2/benchmarks/{bid}", json=fields).raise_for_status()
        except httpx.HTTPError as e:
            logger.warning("benchmark update failed: %s", e)
