"""Worker-topology fixture loaders (reference test strategy: SURVEY.md §4 —
serialized Worker rows with full gpu_devices drive multi-node scheduling
tests without a cluster)."""
from __future__ import annotations

import json
from pathlib import Path

HERE = Path(__file__).resolve().parent


def load_worker(name: str, worker_id: int, state: str = "ready") -> dict:
    data = json.loads((HERE / f"{name}.json").read_text())
    data["id"] = worker_id
    data["state"] = state
    data["heartbeat_time"] = 10**12
    return data


def mi355x_8g(worker_id: int = 1, idx: int = 0) -> dict:
    return load_worker(f"linux_amd_mi355x_288gx8_{idx}", worker_id)


def mi355x_4g_labeled(worker_id: int = 3) -> dict:
    return load_worker("linux_amd_mi355x_288gx4_labeled", worker_id)
