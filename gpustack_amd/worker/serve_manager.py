"""Worker-side instance lifecycle engine
(reference: gpustack/worker/serve_manager.py:184).

Watches ModelInstance events for this worker and drives the state machine:
SCHEDULED -> INITIALIZING -> (DOWNLOADING) -> STARTING -> RUNNING, with
port assignment from the configured range, per-instance log capture,
health probing, and crash restarts with exponential backoff capped at
300 s (reference: serve_manager.py:1842-1885).

Engine processes are subprocesses of this agent (`python -m
gpustack_amd.worker.engine_server`) with HIP_VISIBLE_DEVICES pinned to the
scheduled GPUs — the MI355X-native replacement for the reference's
Docker/K8s workload creation.
"""
from __future__ import annotations

import json
import logging
import os
import socket
import subprocess
import sys
import threading
import time
from pathlib import Path

import httpx

from ..client import ServerClient
from ..config import Config
from ..schemas import ModelInstanceState as S

logger = logging.getLogger(__name__)

RESTART_BASE = 10.0
RESTART_CAP = 300.0


class InstanceProcess:
    def __init__(self, instance: dict, proc: subprocess.Popen, port: int, log_path: Path):
        self.instance = instance
        self.proc = proc
        self.port = port
        self.log_path = log_path
        self.started_at = time.time()
        self.healthy = False
        self.role = "main"
        self.fence = None  # CgroupFence when resource limits are requested


class ServeManager:
    def __init__(self, cfg: Config, client: ServerClient, worker_id: int):
        self.cfg = cfg
        self.client = client
        self.worker_id = worker_id
        self.processes: dict[int, InstanceProcess] = {}  # instance_id -> proc
        self._used_ports: set[int] = set()
        self._lock = threading.Lock()
        self._restart_at: dict[int, float] = {}
        self._stop = False

    # ---- orphan cleanup (reference: worker/workload_cleaner.py — GC of
    # containers a previous worker run left behind) -------------------------

    def cleanup_orphans(self) -> None:
        """Kill engine_server processes a PREVIOUS agent run left behind.

        The DB records each instance's pid; a fresh agent has no entry in
        self.processes for them, so a still-alive pid would double-serve
        the GPU and squat the port. Identity is verified via
        /proc/<pid>/cmdline (exact module + --served-name) before the kill
        — never pattern-matched."""
        import os
        import signal

        try:
            insts = self.client.list_instances(worker_id=self.worker_id)
        except Exception:  # noqa: BLE001
            return
        for inst in insts:
            pid = inst.get("pid")
            if not pid or inst.get("id") in self.processes:
                continue
            try:
                with open(f"/proc/{pid}/cmdline", "rb") as f:
                    argv = f.read().split(b"\0")
            except OSError:
                continue  # not running
            if b"gpustack_amd.worker.engine_server" not in argv:
                continue  # pid was recycled by an unrelated process
            if inst.get("model_name", "").encode() not in argv:
                continue
            logger.warning("killing orphan engine process pid=%s (instance %s "
                           "from a previous agent run)", pid, inst.get("name"))
            try:
                os.killpg(os.getpgid(pid), signal.SIGTERM)
            except (OSError, ProcessLookupError):
                try:
                    os.kill(pid, signal.SIGTERM)
                except (OSError, ProcessLookupError):
                    continue

    # ---- main loops ------------------------------------------------------

    def watch_loop(self) -> None:
        self.cleanup_orphans()
        while not self._stop:
            try:
                for frame in self.client.watch_instances(self.worker_id):
                    if self._stop:
                        return
                    t = frame.get("type")
                    data = frame.get("data", {})
                    if t == "HEARTBEAT":
                        continue
                    role = self._role_for(data)
                    if role is None:
                        if t == "DELETED" and data.get("id") in self.processes:
                            self._stop_instance(data["id"])
                        continue
                    self.dispatch(t, data, role)
            except Exception as e:  # noqa: BLE001
                logger.warning("watch stream broken (%s); reconnecting", e)
                time.sleep(3)

    def health_loop(self) -> None:
        while not self._stop:
            try:
                self.check_health()
            except Exception:  # noqa: BLE001
                logger.exception("health cycle failed")
            time.sleep(5.0)

    def stop(self) -> None:
        self._stop = True
        for iid in list(self.processes):
            self._stop_instance(iid)

    # ---- event dispatch (serve_manager.py:968) ---------------------------

    def _role_for(self, inst: dict) -> str | None:
        """main | sub | None — cross-worker TP instances involve this
        worker either as rank-0 host or as a subordinate rank host
        (reference: distributed_servers / coordinate modes)."""
        if inst.get("worker_id") == self.worker_id:
            return "main"
        ds = inst.get("distributed_servers") or {}
        for sub in ds.get("subordinates", []):
            if sub.get("worker_id") == self.worker_id:
                return "sub"
        return None

    def dispatch(self, event_type: str, inst: dict, role: str = "main") -> None:
        iid = inst["id"]
        state = inst.get("state")
        if event_type == "DELETED":
            self._stop_instance(iid)
            return
        if state == S.SCHEDULED.value and iid not in self.processes:
            try:
                self._start_instance(inst, role)
            except Exception as e:  # noqa: BLE001
                logger.exception("failed to start instance %s", inst.get("name"))
                self._safe_update(iid, state=S.ERROR.value, state_message=str(e))

    # ---- lifecycle -------------------------------------------------------

    LOG_ROTATE_BYTES = 16 << 20
    LOG_KEEP = 2

    def _rotate_log(self, log_path: Path) -> None:
        """Size-based rotation (reference: container-log persistence with
        rotation, serve_manager.py:1131-1518): <name>.log -> .log.1 -> .log.2
        once the live file passes LOG_ROTATE_BYTES."""
        try:
            if not log_path.exists() or log_path.stat().st_size < self.LOG_ROTATE_BYTES:
                return
            for i in range(self.LOG_KEEP, 0, -1):
                src = Path(f"{log_path}.{i - 1}") if i > 1 else log_path
                dst = Path(f"{log_path}.{i}")
                if src.exists():
                    src.replace(dst)
        except OSError as e:
            logger.warning("log rotation failed for %s: %s", log_path, e)

    def _assign_port(self) -> int:
        lo, hi = self.cfg.engine_port_range()
        with self._lock:
            for p in range(lo, hi):
                if p in self._used_ports:
                    continue
                with socket.socket() as s:
                    try:
                        s.bind(("", p))
                    except OSError:
                        continue
                self._used_ports.add(p)
                return p
        raise RuntimeError("no free ports in range")

    def _start_instance(self, inst: dict, role: str = "main") -> None:
        iid = inst["id"]
        model = self.client.get_model(inst["model_id"])
        if role == "main":
            self._safe_update(iid, state=S.INITIALIZING.value)

        source = model.get("source", "preset")
        ref = model["model_ref"]
        if source == "local_path" and not Path(ref).exists():
            self._safe_update(iid, state=S.ERROR.value,
                              state_message=f"local path {ref} not found")
            return
        if source == "huggingface":
            local = self._download_model(iid, ref)
            if local is None:
                return
            source, ref = "local_path", local

        port = self._assign_port()
        if inst.get("distributed_servers"):
            # distributed instances get a fenced port BAND (reference:
            # serve_manager.py:1643-1739): the rendezvous master port and
            # follower slots around it must not be handed to the next
            # instance on this worker
            ds_ports = inst["distributed_servers"]
            band = [ds_ports.get("master_port")]
            with self._lock:
                for p in band:
                    if p:
                        self._used_ports.add(p)
                for fence in range(port + 1, port + 4):
                    self._used_ports.add(fence)
        log_dir = Path(self.cfg.data_dir) / "log" / "instances"
        log_dir.mkdir(parents=True, exist_ok=True)
        log_path = log_dir / f"{inst['name']}.log"
        self._rotate_log(log_path)

        env = dict(os.environ)
        ds = inst.get("distributed_servers") or None
        my_sub = None
        if ds and role == "sub":
            my_sub = next(x for x in ds["subordinates"]
                          if x["worker_id"] == self.worker_id)
        gpus = (my_sub["gpu_indexes"] if my_sub else inst.get("gpu_indexes")) or []
        if gpus:
            env["HIP_VISIBLE_DEVICES"] = ",".join(str(g) for g in gpus)
            env["CUDA_VISIBLE_DEVICES"] = env["HIP_VISIBLE_DEVICES"]
        env.update(model.get("env") or {})

        bp = dict(model.get("backend_parameters") or {})
        bp.pop("cpu_offload", None)  # scheduler knob, not an engine field
        claim = inst.get("computed_resource_claim") or {}
        if claim.get("offload_gb"):
            # scheduler placed this instance with partial CPU offload
            # (engine/offload.py streams the tail layers from host DRAM)
            bp.setdefault("cpu_offload_gb", claim["offload_gb"])
        if claim.get("pp_partition"):
            # uneven pipeline stages (per-GPU tensor_split analog)
            bp.setdefault("pp_partition", claim["pp_partition"])
        if model.get("lora_list"):
            bp.setdefault("lora_dirs", model["lora_list"])
        if model.get("lora_adapters"):
            bp.setdefault("lora_adapters", model["lora_adapters"])
        args = [
            sys.executable, "-m", "gpustack_amd.worker.engine_server",
            "--served-name", model["name"],
            "--source", source,
            "--model-ref", ref,
            "--port", str(port),
            "--gpu-memory-utilization", str(model.get("gpu_memory_utilization") or 0.9),
        ]
        if model.get("max_model_len"):
            args += ["--max-model-len", str(model["max_model_len"])]
        if ds:
            # cross-worker TP: rank group layout from the scheduler
            args += ["--tp", str(ds["tp"]),
                     "--local-ranks", str(len(gpus)),
                     "--rank-base", str(my_sub["rank_base"] if my_sub else 0),
                     "--master-addr", ds["master_ip"] or "127.0.0.1",
                     "--master-port", str(ds["master_port"])]
        elif len(gpus) > 1:
            # TP replica sharded over the scheduled GPUs (RCCL over xGMI);
            # backend_parameters pp_size splits layers into pipeline stages
            # instead (tp x pp = scheduled GPUs)
            pp = int(bp.pop("pp_size", 1) or 1)
            cp = int(bp.pop("cp_size", bp.pop("pcp_size", 1) or 1) or 1)
            if pp > 1 and len(gpus) % pp == 0:
                args += ["--tp", str(len(gpus) // pp), "--pp", str(pp)]
            elif cp > 1 and len(gpus) % cp == 0:
                # prefill context parallelism: tp x cp = scheduled GPUs
                args += ["--tp", str(len(gpus) // cp), "--cp", str(cp)]
            else:
                args += ["--tp", str(len(gpus))]
        if bp:
            args += ["--backend-parameters", json.dumps(bp)]

        # resource fencing (reference: gpustack-runtime container limits):
        # backend_parameters memory_limit_gb / cpu_limit / pids_limit put
        # the engine in its own cgroup — best-effort, never blocks serving
        from .isolation import fence_from_backend_parameters

        fence = fence_from_backend_parameters(
            f"{inst['id']}-{role}", model.get("backend_parameters") or {})
        logf = open(log_path, "ab")
        proc = subprocess.Popen(args, env=env, stdout=logf, stderr=subprocess.STDOUT,
                                start_new_session=True)
        if fence is not None and fence.create():
            fence.attach(proc.pid)
        ip = InstanceProcess(inst, proc, port, log_path)
        ip.fence = fence
        ip.role = role
        self.processes[iid] = ip
        if role == "main":
            self._safe_update(iid, state=S.STARTING.value, port=port, pid=proc.pid)
        logger.info("instance %s starting (%s): pid=%d port=%d gpus=%s",
                    inst["name"], role, proc.pid, port, gpus)

    def _download_model(self, iid: int, repo: str) -> str | None:
        """Resolve HF files under an OS file lock (reference: model-file
        manager download locks, model_file_manager.py:375-427 — two replicas
        of one model must not download concurrently) and record a ModelFile
        row so the scheduler's locality scorer can prefer this worker."""
        import fcntl
        import hashlib

        self._safe_update(iid, state=S.DOWNLOADING.value)
        try:
            from huggingface_hub import snapshot_download

            cache = Path(self.cfg.cache_dir or Path(self.cfg.data_dir) / "cache")
            cache.mkdir(parents=True, exist_ok=True)
            lock_path = cache / f".{hashlib.sha256(repo.encode()).hexdigest()[:16]}.lock"
            with open(lock_path, "w") as lf:
                fcntl.flock(lf, fcntl.LOCK_EX)
                try:
                    path = snapshot_download(repo, cache_dir=str(cache))
                finally:
                    fcntl.flock(lf, fcntl.LOCK_UN)
            self._record_model_file("huggingface", repo, path)
            return path
        except Exception as e:  # noqa: BLE001
            self._safe_update(iid, state=S.ERROR.value,
                              state_message=f"download failed: {e}")
            return None

    def _record_model_file(self, source: str, ref: str, path: str) -> None:
        try:
            size = sum(f.stat().st_size for f in Path(path).rglob("*")
                       if f.is_file())
            self.client._c.post("/v2/model_files", json={
                "worker_id": self.worker_id, "source": source,
                "model_ref": ref, "local_path": str(path),
                "size_bytes": size, "state": "ready",
            }).raise_for_status()
        except Exception as e:  # noqa: BLE001
            logger.warning("model-file record failed: %s", e)

    def _stop_instance(self, iid: int) -> None:
        ip = self.processes.pop(iid, None)
        self._restart_at.pop(iid, None)
        if ip is None:
            return
        logger.info("stopping instance %s (pid %s)", ip.instance.get("name"), ip.proc.pid)
        try:
            os.killpg(ip.proc.pid, 15)
        except ProcessLookupError:
            pass
        try:
            ip.proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            try:
                os.killpg(ip.proc.pid, 9)
            except ProcessLookupError:
                pass
        if ip.fence is not None:
            ip.fence.cleanup()
        with self._lock:
            self._used_ports.discard(ip.port)

    # ---- health / restart ------------------------------------------------

    def check_health(self) -> None:
        for iid, ip in list(self.processes.items()):
            rc = ip.proc.poll()
            if rc is not None:
                self._handle_exit(iid, ip, rc)
                continue
            if getattr(ip, "role", "main") == "sub":
                continue  # subordinate rank hosts have no HTTP endpoint
            if not ip.healthy:
                try:
                    r = httpx.get(f"http://127.0.0.1:{ip.port}/health", timeout=3.0)
                    if r.status_code == 200:
                        ip.healthy = True
                        self._safe_update(iid, state=S.RUNNING.value, state_message="")
                        logger.info("instance %s RUNNING", ip.instance.get("name"))
                except httpx.HTTPError:
                    if time.time() - ip.started_at > 600:
                        self._safe_update(iid, state=S.ERROR.value,
                                          state_message="startup timeout")
                        self._stop_instance(iid)

    def _handle_exit(self, iid: int, ip: InstanceProcess, rc: int) -> None:
        self.processes.pop(iid, None)
        with self._lock:
            self._used_ports.discard(ip.port)
        inst = ip.instance
        if getattr(ip, "role", "main") == "sub":
            # a dead subordinate stalls the TP group; surface the error and
            # leave recovery to delete/recreate (round-2: group restart)
            self._safe_update(iid, state=S.ERROR.value,
                              state_message=f"subordinate rank host exited {rc}")
            return
        restart_count = (inst.get("restart_count") or 0) + 1
        model = None
        try:
            model = self.client.get_model(inst["model_id"])
        except httpx.HTTPError:
            pass
        restart = (model or {}).get("restart_on_error", True)
        msg = f"engine exited with code {rc}"
        logger.warning("instance %s: %s", inst.get("name"), msg)
        if restart:
            delay = min(RESTART_CAP, RESTART_BASE * 2 ** min(restart_count - 1, 8))
            self._safe_update(iid, state=S.ERROR.value, state_message=msg,
                              restart_count=restart_count)
            inst2 = dict(inst, restart_count=restart_count)
            self._restart_at[iid] = time.time() + delay
            threading.Timer(delay, self._try_restart, args=(iid, inst2)).start()
        else:
            self._safe_update(iid, state=S.ERROR.value, state_message=msg)

    def _try_restart(self, iid: int, inst: dict) -> None:
        if self._stop or iid in self.processes:
            return
        try:
            self._start_instance(inst)
        except Exception as e:  # noqa: BLE001
            self._safe_update(iid, state=S.ERROR.value, state_message=str(e))

    def _safe_update(self, iid: int, **fields) -> None:
        try:
            self.client.update_instance(iid, **fields)
        except httpx.HTTPError as e:
            logger.warning("instance %d update failed: %s", iid, e)
