"""TTL cache + stampede protection + DB-generation invalidation
(reference server/cache.py re-designed for DB-lease HA)."""
import threading
import time


def _setup(tmp_path):
    from gpustack_amd.db import init_db

    init_db(f"sqlite:///{tmp_path}/c.db")


def test_ttl_and_lru(tmp_path):
    _setup(tmp_path)
    from gpustack_amd.server.cache import TTLCache

    c = TTLCache(ttl=0.05, maxsize=2)
    c.set("a", 1)
    assert c.get("a") == 1
    time.sleep(0.06)
    assert c.get("a") is None
    c.set("a", 1); c.set("b", 2); c.set("c", 3)
    assert len(c) == 2 and c.get("a") is None  # LRU evicted


def test_locked_cached_stampede(tmp_path):
    _setup(tmp_path)
    from gpustack_amd.server.cache import TTLCache, locked_cached

    calls = []

    @locked_cached(TTLCache(ttl=10))
    def slow(x):
        calls.append(x)
        time.sleep(0.05)
        return x * 2

    threads = [threading.Thread(target=slow, args=(7,)) for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert slow(7) == 14
    assert len(calls) == 1  # computed exactly once despite 8 racers


def test_distributed_invalidation(tmp_path):
    _setup(tmp_path)
    from gpustack_amd.server.cache import TTLCache

    # two caches simulating two server instances sharing the DB
    a = TTLCache(ttl=60, distributed_name="t", check_interval=0.0)
    b = TTLCache(ttl=60, distributed_name="t", check_interval=0.0)
    a.set("k", "v1")
    b.set("k", "v1")
    assert b.get("k") == "v1"
    a.invalidate()  # bumps the DB generation
    assert a.get("k") is None
    assert b.get("k") is None  # b sees the bump via the DB


def test_eval_cache_integration(tmp_path):
    # model-evaluations result served from cache on repeat call
    import tempfile

    from fastapi.testclient import TestClient

    import gpustack_amd.server.routes_v2 as rv2
    from gpustack_amd.config import Config
    from gpustack_amd.server.app import create_app

    rv2._eval_cache = None
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123")
    app = create_app(cfg, start_background=False)
    c = TestClient(app)
    tok = c.post("/auth/login", json={"username": "admin",
                                      "password": "pw123"}).json()["token"]
    c.headers["Authorization"] = f"Bearer {tok}"
    body = {"name": "ev", "model_ref": "tiny", "source": "preset",
            "gpus_per_replica": 1}
    r1 = c.post("/v2/model-evaluations", json=body)
    assert r1.status_code == 200
    cache = rv2.get_eval_cache()
    n = len(cache)
    r2 = c.post("/v2/model-evaluations", json=body)
    assert r2.json() == r1.json() and len(cache) == n
    rv2._eval_cache = None
