"""Tunnel proxy e2e: worker in NAT mode serves a model; inference traffic
flows through the worker-initiated long-poll tunnel, not a direct
connection to the engine port."""
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
def test_tunnel_mode_serving():
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

    lo = _free_port()
    wcfg = Config(
        data_dir=tempfile.mkdtemp(), server_url=base,
        token=app.state.bootstrap["registration_token"],
        worker_name="nat-worker", worker_ip="203.0.113.7",  # unroutable IP
        worker_port=_free_port(), port_range=f"{lo}-{lo + 50}",
        proxy_mode="tunnel",
        gpu_devices=[{"index": 0, "name": "AMD Instinct MI355X",
                      "memory": {"total": 288 * 1024**3}}],
        heartbeat_interval=2.0, worker_status_interval=5.0,
    )
    agent = WorkerAgent(wcfg)
    threading.Thread(target=agent.start, daemon=True).start()

    client = httpx.Client(base_url=base, timeout=60)
    tok = client.post("/auth/login", json={"username": "admin", "password": "pw"}).json()["token"]
    client.headers["Authorization"] = f"Bearer {tok}"
    try:
        for _ in range(100):
            ws = client.get("/v2/workers").json()["items"]
            if ws:
                assert ws[0]["proxy_mode"] == "tunnel"
                break
            time.sleep(0.2)

        r = client.post("/v2/models", json={
            "name": "tiny-nat", "source": "preset", "model_ref": "tiny",
            "replicas": 1, "max_model_len": 256})
        assert r.status_code == 201
        state = None
        for _ in range(240):
            insts = client.get("/v2/model_instances").json()["items"]
            if insts:
                state = insts[0]["state"]
                if state == "running":
                    break
                assert state != "error", insts[0]["state_message"]
            time.sleep(0.5)
        assert state == "running"

        # the registered worker IP is unroutable: the ONLY way the proxy can
        # answer is through the tunnel
        r = client.post("/v1/chat/completions", json={
            "model": "tiny-nat", "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 4, "ignore_eos": True})
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 4
        # streaming through the tunnel
        with client.stream("POST", "/v1/chat/completions", json={
            "model": "tiny-nat", "messages": [{"role": "user", "content": "s"}],
            "max_tokens": 3, "stream": True, "ignore_eos": True}) as resp:
            assert resp.status_code == 200
            frames = [l for l in resp.iter_lines() if l.startswith("data:")]
        assert frames and frames[-1].strip() == "data: [DONE]"

        # CONCURRENT streams through the tunnel (r2 batched job pickup +
        # per-request reply channels): 8 simultaneous streaming requests
        # must all complete with their own full frame sequences
        import concurrent.futures

        def one_stream(i: int) -> tuple[int, int, bool]:
            c2 = httpx.Client(base_url=base, timeout=60)
            c2.headers["Authorization"] = client.headers["Authorization"]
            with c2.stream("POST", "/v1/chat/completions", json={
                "model": "tiny-nat",
                "messages": [{"role": "user", "content": f"req {i}"}],
                "max_tokens": 4, "stream": True,
                "ignore_eos": True}) as resp:
                fl = [l for l in resp.iter_lines() if l.startswith("data:")]
            c2.close()
            return (resp.status_code, len(fl),
                    bool(fl) and fl[-1].strip() == "data: [DONE]")

        with concurrent.futures.ThreadPoolExecutor(8) as pool:
            results = list(pool.map(one_stream, range(8)))
        assert all(st == 200 for st, _, _ in results), results
        assert all(done for _, _, done in results), results
        # each stream carried events of its own (frames may coalesce
        # under concurrency; completion + [DONE] is the real contract)
        assert all(n >= 2 for _, n, _ in results), results
    finally:
        stop_background_tasks(app)
        agent.stop()
        server.should_exit = True
