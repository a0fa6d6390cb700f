"""Driver contract: bench.py must emit ONE valid JSON line, including
under torch.distributed.run with N ranks (the driver's scale runs)."""
import json
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_bench_single_rank_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--device", "cpu", "--model", "tiny",
         "--steps", "2", "--warmup", "1", "--concurrency", "2",
         "--isl", "8", "--osl", "4", "--max-model-len", "64"],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")][-1]
    j = json.loads(line)
    assert j["n_gpus"] == 1 and j["value"] > 0
    for key in ("metric", "unit", "ms_per_step", "higher_is_better",
                "scaling", "vs_baseline", "dtype", "data", "config"):
        assert key in j


@pytest.mark.timeout(600)
def test_bench_two_ranks_gloo():
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2",
         "--device", "cpu", "--model", "tiny", "--steps", "2",
         "--warmup", "1", "--concurrency", "2", "--isl", "8", "--osl", "4",
         "--max-model-len", "64"],
        capture_output=True, text=True, timeout=540)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith('{"metric"')]
    assert len(lines) == 1, f"exactly rank0 prints: {lines}"
    j = json.loads(lines[0])
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
