"""Placement policy tests against MI355X worker-topology fixtures
(mirrors the reference's fixture-driven scheduler tests,
tests/policies/candidate_selectors in the reference)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

from fixtures.workers.fixtures import mi355x_4g_labeled, mi355x_8g

from gpustack_amd.scheduler.policies import (
    estimate_vram_claim,
    model_spec_for,
    pick_candidate,
    select_candidates,
    worker_allocatable,
)


def _model(**kw):
    base = dict(id=1, name="m", source="preset", model_ref="llama-3-8b",
                replicas=1, gpus_per_replica=1, gpu_memory_utilization=0.9,
                placement_strategy="binpack", worker_selector={}, gpu_selector=None)
    base.update(kw)
    return base


def test_memory_model_llama8b():
    spec = model_spec_for(_model())
    assert spec is not None
    claim = estimate_vram_claim(_model(), spec, tp=1)
    # 8B bf16 weights ~16G * 1.2 + 2G overhead + 4G KV floor ≈ 25-26 GiB
    assert 20 << 30 < claim < 32 << 30


def test_allocatable_subtracts_claims_and_reserved():
    w = mi355x_8g(1)
    inst = {"worker_id": 1, "computed_resource_claim": {"vram": {"0": 100 << 30}}}
    alloc = worker_allocatable(w, [inst])
    total = 288 * 1024**3
    assert alloc[0] == total - (1 << 30) - (100 << 30)
    assert alloc[1] == total - (1 << 30)


def test_pick_single_gpu_candidate():
    w = mi355x_8g(1)
    cand = pick_candidate(_model(), [w], [])
    assert cand is not None
    assert len(cand.gpu_indexes) == 1


def test_tp8_uses_all_gpus():
    w = mi355x_8g(1)
    cand = pick_candidate(_model(model_ref="llama-3-70b", gpus_per_replica=8), [w], [])
    assert cand is not None
    assert sorted(cand.gpu_indexes) == list(range(8))


def test_binpack_prefers_busier_worker():
    w1, w2 = mi355x_8g(1, 0), mi355x_8g(2, 1)
    # one existing instance on w1
    insts = [{"worker_id": 1, "model_id": 99,
              "computed_resource_claim": {"vram": {"0": 30 << 30}}}]
    cand = pick_candidate(_model(placement_strategy="binpack"), [w1, w2], insts)
    assert cand.worker["id"] == 1


def test_spread_prefers_empty_worker():
    w1, w2 = mi355x_8g(1, 0), mi355x_8g(2, 1)
    insts = [{"worker_id": 1, "model_id": 99,
              "computed_resource_claim": {"vram": {"0": 30 << 30}}}]
    cand = pick_candidate(_model(placement_strategy="spread"), [w1, w2], insts)
    assert cand.worker["id"] == 2


def test_label_selector_filters():
    w1, w3 = mi355x_8g(1), mi355x_4g_labeled(3)
    cand = pick_candidate(_model(worker_selector={"pool": "small"}), [w1, w3], [])
    assert cand.worker["id"] == 3


def test_manual_gpu_selection():
    w = mi355x_8g(1)
    cand = pick_candidate(
        _model(gpu_selector={"gpu_ids": ["1:5", "1:6"]}, gpus_per_replica=2),
        [w], [])
    assert cand.gpu_indexes == [5, 6]


def test_not_ready_workers_filtered():
    w = mi355x_8g(1)
    w["state"] = "not_ready"
    assert pick_candidate(_model(), [w], []) is None


def test_full_worker_rejected():
    w = mi355x_8g(1)
    # fill every GPU with huge claims
    insts = [{"worker_id": 1, "computed_resource_claim":
              {"vram": {str(i): 280 << 30 for i in range(8)}}}]
    assert pick_candidate(_model(), [w], insts) is None


def test_replicas_spread_across_gpus():
    """8 replicas on one 8-GPU node land on 8 distinct GPUs (the bench
    deployment shape: BASELINE.json config 3)."""
    w = mi355x_8g(1)
    placed = []
    insts = []
    for i in range(8):
        cand = pick_candidate(_model(id=1), [w], insts)
        assert cand is not None
        gpu = cand.gpu_indexes[0]
        placed.append(gpu)
        insts.append({"worker_id": 1, "model_id": 1,
                      "computed_resource_claim": {"vram": {str(gpu): c}}}
                     if (c := estimate_vram_claim(_model(), model_spec_for(_model()), 1))
                     else {})
    assert sorted(placed) == list(range(8))


def test_file_locality_scorer_prefers_holder():
    """ModelFileLocalityScorer analog: at equal fit, the worker already
    holding the checkpoint wins placement."""
    import sys
    sys.path.insert(0, "tests")
    from fixtures.workers.fixtures import mi355x_8g

    from gpustack_amd.scheduler.policies import pick_candidate

    w1, w2 = mi355x_8g(1), mi355x_8g(2)
    w1["id"], w1["name"] = 1, "w1"
    w2["id"], w2["name"] = 2, "w2"
    w1["state"] = w2["state"] = "ready"
    model = {"id": 9, "name": "m", "source": "huggingface",
             "model_ref": "org/llama", "gpus_per_replica": 1,
             "placement_strategy": "spread",
             "gpu_memory_utilization": 0.9, "max_model_len": 2048,
             "categories": ["llm"]}
    files = [{"worker_id": 2, "source": "huggingface",
              "model_ref": "org/llama", "state": "ready"}]
    cand = pick_candidate(model, [w1, w2], [], files)
    assert cand.worker["id"] == 2
    # without the file record the tie breaks by id order (w1)
    cand = pick_candidate(model, [w1, w2], [], [])
    assert cand.worker["id"] == 1


def test_gpu_type_selector_filters_devices():
    """Device-class placement (reference gpu_type_selector): only devices
    matching the partition/VRAM class are eligible."""
    from gpustack_amd.scheduler.policies import (device_matches_type,
                                                 select_candidates)

    big = {"index": 0, "type": "rocm", "name": "AMD Instinct MI355X",
           "memory": {"total": 288 << 30},
           "partition": {"compute": "SPX", "memory": "NPS1"}}
    slice_ = {"index": 1, "type": "rocm", "name": "AMD Instinct MI355X",
              "memory": {"total": 36 << 30},
              "partition": {"compute": "CPX", "memory": "NPS1"}}
    assert device_matches_type(big, {"partition_compute": "SPX"})
    assert not device_matches_type(slice_, {"partition_compute": "SPX"})
    assert device_matches_type(slice_, {"min_vram_gb": 32})
    assert not device_matches_type(slice_, {"min_vram_gb": 64})
    assert device_matches_type(big, {"name_contains": "mi355"})

    worker = {"id": 1, "name": "w", "state": "ready",
              "status": {"gpu_devices": [big, slice_]},
              "system_reserved": {}}
    model = {"id": 1, "source": "preset", "model_ref": "tiny",
             "gpus_per_replica": 1,
             "gpu_type_selector": {"partition_compute": "CPX"}}
    cands = select_candidates(model, [worker], [])
    assert len(cands) == 1 and cands[0].gpu_indexes == [1]
    model["gpu_type_selector"] = {"partition_compute": "SPX"}
    cands = select_candidates(model, [worker], [])
    assert cands and cands[0].gpu_indexes == [0]
