"""Cross-worker tensor parallelism end-to-end on CPU (gloo).

The reference orchestrates multi-worker inference via subordinate workers
(SURVEY.md §2.10); here the whole path is first-party: multi-worker
placement -> distributed_servers rank layout -> each worker's serve
manager hosts its rank group -> TP over torch.distributed -> OpenAI
completion through the gateway.
"""
import socket
import tempfile
import threading
import time

import httpx
import pytest


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.mark.timeout(300)
def test_cross_worker_tp2():
    import uvicorn

    from gpustack_amd.config import Config
    from gpustack_amd.server.app import create_app, stop_background_tasks
    from gpustack_amd.worker.agent import WorkerAgent

    sport = _free_port()
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw",
                 host="127.0.0.1", port=sport)
    app = create_app(cfg, start_background=True)
    server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=sport,
                                           log_level="warning"))
    threading.Thread(target=server.run, daemon=True).start()
    base = f"http://127.0.0.1:{sport}"
    for _ in range(100):
        try:
            httpx.get(base + "/healthz", timeout=1)
            break
        except httpx.HTTPError:
            time.sleep(0.1)

    agents = []
    for i in range(2):
        lo = _free_port()
        wcfg = Config(
            data_dir=tempfile.mkdtemp(), server_url=base,
            token=app.state.bootstrap["registration_token"],
            worker_name=f"w{i}", worker_ip="127.0.0.1",
            worker_port=_free_port(), port_range=f"{lo}-{lo + 50}",
            gpu_devices=[{"index": 0, "name": "AMD Instinct MI355X",
                          "memory": {"total": 288 * 1024**3}}],
            heartbeat_interval=2.0, worker_status_interval=5.0,
        )
        a = WorkerAgent(wcfg)
        threading.Thread(target=a.start, daemon=True).start()
        agents.append(a)

    client = httpx.Client(base_url=base, timeout=30)
    tok = client.post("/auth/login", json={"username": "admin", "password": "pw"}).json()["token"]
    client.headers["Authorization"] = f"Bearer {tok}"
    for _ in range(150):
        if len(client.get("/v2/workers").json()["items"]) == 2:
            break
        time.sleep(0.2)

    try:
        r = client.post("/v2/models", json={
            "name": "tiny-dist", "source": "preset", "model_ref": "tiny",
            "replicas": 1, "max_model_len": 256, "gpus_per_replica": 2,
            "distributed_inference_across_workers": True,
        })
        assert r.status_code == 201, r.text

        state = None
        inst = None
        for _ in range(480):  # generous: full-suite CPU load slows gloo rendezvous
            insts = client.get("/v2/model_instances").json()["items"]
            if insts:
                inst = insts[0]
                state = inst["state"]
                if state == "running":
                    break
                assert state != "error", inst["state_message"]
            time.sleep(0.5)
        assert state == "running", f"never ran (state={state})"
        ds = inst["distributed_servers"]
        assert ds and ds["tp"] == 2 and len(ds["subordinates"]) == 1
        assert ds["subordinates"][0]["worker_id"] != inst["worker_id"]
        # both workers host a rank process
        hosts = sum(1 for a in agents if a.serve_manager.processes)
        assert hosts == 2

        r = client.post("/v1/completions", json={
            "model": "tiny-dist", "prompt": "abc", "max_tokens": 5,
            "ignore_eos": True, "temperature": 0,
        })
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 5
    finally:
        stop_background_tasks(app)
        for a in agents:
            a.stop()
        server.should_exit = True


def test_orphan_cleanup_kills_stale_engine_process(tmp_path):
    """A fresh agent kills engine_server pids a previous run left behind
    (identity verified via /proc cmdline — reference WorkloadCleaner)."""
    import os
    import signal
    import subprocess
    import sys
    import time as _t

    from gpustack_amd.config import Config
    from gpustack_amd.worker.serve_manager import ServeManager

    # a real orphan: engine_server for model "tiny-orphan" with no agent
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-orphan", "--source", "preset",
        "--model-ref", "tiny", "--port", "0", "--device", "cpu",
        "--kv-cache-blocks", "32", "--max-model-len", "128",
    ], start_new_session=True)
    try:
        class _FakeClient:
            def list_instances(self, worker_id=None):
                return [{"id": 1, "pid": proc.pid, "model_name": "tiny-orphan",
                         "name": "tiny-orphan-0"},
                        {"id": 2, "pid": 999999, "model_name": "x",
                         "name": "gone"},  # dead pid: skipped
                        {"id": 3, "pid": os.getppid(), "model_name": "x",
                         "name": "unrelated"}]  # live but wrong cmdline: kept

        sm = ServeManager(Config(data_dir=str(tmp_path)), _FakeClient(), 1)
        sm.cleanup_orphans()
        for _ in range(100):
            if proc.poll() is not None:
                break
            _t.sleep(0.1)
        assert proc.poll() is not None  # orphan terminated
        # the unrelated pid (our parent) is untouched
        os.kill(os.getppid(), 0)
    finally:
        if proc.poll() is None:
            os.killpg(os.getpgid(proc.pid), signal.SIGKILL)
