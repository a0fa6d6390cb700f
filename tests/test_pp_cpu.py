"""Pipeline-parallel engine correctness on CPU (gloo, world_size 2).

PP=2 stage-partitioned layers with one hidden-state handoff per step must
produce exactly the same greedy continuation as the single-rank engine:
the boundary sends the summed residual stream in bf16, which is the same
value fused_add_rms_norm would have stored (reference: vLLM
--pipeline-parallel-size, SURVEY.md §2.10 PP row)."""
import json
import multiprocessing as mp
import os
import socket
import tempfile

import pytest


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


PROMPTS = [[3, 1, 4, 1, 5, 9, 2, 6], [11, 22, 33]]


def _single_proc_result(model: str = "tiny", **params) -> list[list[int]]:
    import torch

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    # single-thread math: CPU-parallel reductions reorder under load, and
    # 1-ulp drift flips the MoE router's top-k (documented fragility) —
    # both sides of the exactness comparison must reduce in the same order
    torch.set_num_threads(1)

    cfg = EngineConfig(model=model, device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0)
    eng = LLMEngine(cfg)
    return eng.generate(PROMPTS, SamplingParams(max_tokens=6, ignore_eos=True,
                                                **params))


def _pp_rank_main(rank: int, pp: int, port: int, out_path: str,
                  model: str, params: dict):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_parallel

    torch.set_num_threads(1)  # match _single_proc_result's reduction order

    comm = init_parallel(1, pp, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model=model, device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0)
    eng = LLMEngine(cfg, comm)
    # stage partition sanity
    assert eng.runner.model.num_local_layers >= 1
    assert eng.scheduler.kv.num_layers == eng.runner.model.num_local_layers
    results: dict[str, list[int]] = {}
    rids = []
    if rank == 0:
        rids = [eng.add_request(p, SamplingParams(max_tokens=6,
                                                  ignore_eos=True, **params))
                for p in PROMPTS]
        results = {r: [] for r in rids}
    while eng.tp_active():
        outs = eng.step()
        if rank == 0:
            for o in outs:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def _run_pp(pp: int, model: str = "tiny", **params) -> list[list[int]]:
    port = _free_port()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_pp_rank_main,
                         args=(r, pp, port, out_path, model, params))
             for r in range(pp)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0, f"rank process exited {p.exitcode}"
    with open(out_path) as f:
        return json.load(f)


@pytest.mark.timeout(300)
def test_pp2_matches_single_rank():
    assert _run_pp(2) == _single_proc_result()


@pytest.mark.timeout(300)
def test_pp2_sampled_seeded():
    """Seeded temperature sampling happens on the LAST stage; outputs still
    deterministic and relayed to rank 0."""
    out = _run_pp(2, temperature=0.8, seed=12)
    assert all(len(o) == 6 for o in out)
    assert out == _run_pp(2, temperature=0.8, seed=12)


def _pp_ngram_rank_main(rank: int, port: int, out_path: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_parallel

    comm = init_parallel(1, 2, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0,
                       speculative={"method": "ngram",
                                    "num_draft_tokens": 2})
    eng = LLMEngine(cfg, comm)
    results, rids = {}, []
    if rank == 0:
        rids = [eng.add_request(p, SamplingParams(max_tokens=6,
                                                  ignore_eos=True))
                for p in PROMPTS]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_pp2_ngram_spec_matches_plain():
    """ngram speculative drafts verify on the last stage; exactness holds."""
    plain = _single_proc_result()
    port = _free_port()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_pp_ngram_rank_main, args=(r, port, out_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out_path) as f:
        assert json.load(f) == plain


@pytest.mark.timeout(300)
def test_engine_server_pp2_cpu():
    """pp_size path through the engine server: rank 0 spawns the second
    stage, hidden states flow over gloo, OpenAI endpoint answers."""
    import subprocess
    import sys
    import time

    import httpx

    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-pp", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256", "--tp", "1", "--pp", "2",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        t0 = time.time()
        while time.time() - t0 < 120:
            if proc.poll() is not None:
                raise AssertionError(f"engine server exited {proc.returncode}")
            try:
                if httpx.get(f"http://127.0.0.1:{port}/health",
                             timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                pass
            time.sleep(0.5)
        else:
            raise AssertionError("engine server never became healthy")
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-pp", "prompt": "ab", "max_tokens": 6,
            "ignore_eos": True, "temperature": 0,
        }, timeout=120)
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 6
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(300)
def test_pp2_chunked_prefill_matches():
    """Chunked prefill under PP: scheduler state is replicated, chunk and
    decode steps alternate identically on both stages."""
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    long_prompt = [(11 * t + 3) % 500 for t in range(100)]
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, max_prefill_tokens=24,
                       enable_chunked_prefill=True)
    # prompt is clamped to max_model_len-1 internally; compare like for like
    plain = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                   kv_cache_blocks=64, max_model_len=128,
                                   seed=0)).generate(
        [long_prompt], SamplingParams(max_tokens=6, ignore_eos=True))

    port = _free_port()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_pp_chunked_rank_main,
                         args=(r, port, out_path, long_prompt))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out_path) as f:
        assert json.load(f) == plain


def _pp_chunked_rank_main(rank: int, port: int, out_path: str, prompt):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_parallel

    comm = init_parallel(1, 2, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, max_prefill_tokens=24,
                       enable_chunked_prefill=True)
    eng = LLMEngine(cfg, comm)
    results, rids = {}, []
    if rank == 0:
        rids = [eng.add_request(prompt, SamplingParams(max_tokens=6,
                                                       ignore_eos=True))]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def _tpxpp_rank_main(rank: int, port: int, out_path: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_parallel

    comm = init_parallel(2, 2, rank, master_port=port, backend="gloo")
    assert comm.world_size == 4 and comm.world_rank == rank
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, tp_size=2,
                       tp_rank=comm.tp_rank)
    eng = LLMEngine(cfg, comm)
    assert eng.runner.model.num_local_layers == 1  # 2 layers over 2 stages
    results, rids = {}, []
    if rank == 0:
        rids = [eng.add_request(p, SamplingParams(max_tokens=6,
                                                  ignore_eos=True))
                for p in PROMPTS]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp2xpp2_matches_single_rank():
    """Combined TPxPP (world 4, tp-contiguous layout): sharded weights per
    stage + residual handoff reproduce the single-rank outputs exactly."""
    plain = _single_proc_result()
    port = _free_port()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_tpxpp_rank_main, args=(r, port, out_path))
             for r in range(4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0, f"rank exited {p.exitcode}"
    with open(out_path) as f:
        assert json.load(f) == plain


def _pp_lora_rank_main(rank: int, port: int, out_path: str, adapter: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_parallel

    comm = init_parallel(1, 2, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0)
    eng = LLMEngine(cfg, comm)
    results, rids = {}, []
    if rank == 0:
        eng.add_lora("t", adapter)  # replicated to the other stage via ops
        rids = [eng.add_request(PROMPTS[0],
                                SamplingParams(max_tokens=6, ignore_eos=True,
                                               lora_name="t"))]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_pp2_dynamic_lora_matches_single():
    """Dynamic LoRA under PP: the adapter op replicates to every stage and
    each stage applies only its local layers' deltas."""
    import sys
    sys.path.insert(0, "tests")
    from test_lora_dynamic import _make_adapter

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    tmp = tempfile.mkdtemp()
    from pathlib import Path

    _make_adapter(Path(tmp), EngineConfig(model="tiny").spec)
    single = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                    kv_cache_blocks=64, max_model_len=128,
                                    seed=0))
    single.add_lora("t", tmp)
    want = single.generate([PROMPTS[0]],
                           SamplingParams(max_tokens=6, ignore_eos=True,
                                          lora_name="t"))

    port = _free_port()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_pp_lora_rank_main,
                         args=(r, port, out_path, tmp)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out_path) as f:
        assert json.load(f) == want


@pytest.mark.timeout(300)
def test_pp2_moe_matches_single_rank():
    """MoE layers partition across pipeline stages like dense ones."""
    assert _run_pp(2, model="tiny-moe") == _single_proc_result("tiny-moe")


def _pp_part_rank_main(rank: int, pp: int, port: int, out_path: str,
                       partition):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_parallel

    torch.set_num_threads(1)
    comm = init_parallel(1, pp, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, pp_partition=partition)
    cfg.spec.num_layers = 4  # tiny has 2; uneven [3,1] needs 4
    eng = LLMEngine(cfg, comm)
    if partition:
        assert eng.runner.model.num_local_layers == partition[rank]
    results, rids = {}, []
    if rank == 0:
        rids = [eng.add_request(p, SamplingParams(max_tokens=6,
                                                  ignore_eos=True))
                for p in PROMPTS]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def _run_pp_part(pp: int, partition) -> list[list[int]]:
    port = _free_port()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_pp_part_rank_main,
                         args=(r, pp, port, out_path, partition))
             for r in range(pp)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    with open(out_path) as f:
        return json.load(f)


def test_pp_partition_uneven_exact():
    """Uneven per-stage layer counts (pp_partition, the per-GPU
    tensor_split analog) produce output identical to the even split."""
    out_even = _run_pp_part(2, None)
    out_uneven = _run_pp_part(2, [3, 1])
    assert out_even == out_uneven


def test_scheduler_pp_partition_proportional():
    from gpustack_amd.scheduler.policies import Candidate
    from gpustack_amd.scheduler.scheduler import PlacementScheduler
    from gpustack_amd.engine.config import PRESETS

    spec = PRESETS["llama-3-8b"]  # 32 layers
    worker = {"id": 1, "status": {"gpu_devices": [
        {"index": 0, "memory": {"total": 100 << 30}},
        {"index": 1, "memory": {"total": 300 << 30}},
    ]}, "system_reserved": {}}
    cand = Candidate(worker, [0, 1])
    model_d = {"backend_parameters": {"pp_size": 2}}
    part = PlacementScheduler._pp_partition(model_d, spec, cand, [worker], [])
    assert part is not None and sum(part) == 32
    assert part[1] > part[0]  # roomier stage holds more layers
    # even headroom -> engine default (None)
    worker2 = {"id": 1, "status": {"gpu_devices": [
        {"index": 0, "memory": {"total": 200 << 30}},
        {"index": 1, "memory": {"total": 200 << 30}},
    ]}, "system_reserved": {}}
    assert PlacementScheduler._pp_partition(
        model_d, spec, Candidate(worker2, [0, 1]), [worker2], []) is None


@pytest.mark.timeout(300)
def test_pp2_sandwich_norm_model_matches_single_rank():
    """Gemma-class sandwich-norm layers carry the TRUE hidden stream
    (residual sentinel None) — the PP handoff sends x directly instead of
    summing x+residual; output must equal single-rank exactly."""
    expected = _single_proc_result("tiny-gemma")
    got = _run_pp(2, model="tiny-gemma")
    assert got == expected, f"{got} != {expected}"
