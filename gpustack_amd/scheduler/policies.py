"""Placement policies (reference: gpustack/policies/*).

Filter chain -> MI355X resource-fit selector -> scorer chain, with a
claim-based allocation model: allocatable VRAM = device total - sum of
DB-recorded claims - system reserved (reference: policies/utils.py:91-150),
and an analytic VRAM claim for the native engine:
weights x 1.2 + framework overhead + KV pool share
(reference memory model: policies/utils.py:384-470 estimate_model_vram,
re-derived for the first-party engine on 288 GB HBM3E devices).
"""
from __future__ import annotations

import logging
from dataclasses import dataclass, field

from ..engine.config import PRESETS, ModelSpec
from ..schemas import Model, ModelInstance, PlacementStrategy, Worker, WorkerState

logger = logging.getLogger(__name__)

FRAMEWORK_OVERHEAD = 2 << 30      # engine runtime + activations (LLM)
WEIGHT_FUDGE = 1.2
MIN_KV_BYTES = 4 << 30            # refuse placements with <4 GiB KV pool


@dataclass
class Candidate:
    worker: dict
    gpu_indexes: list[int]
    score: float = 0.0
    vram_claim: dict[int, int] = field(default_factory=dict)  # gpu idx -> bytes
    # multi-worker TP (reference: subordinate workers,
    # vllm_resource_fit_selector.py:800-867): [(worker_dict, gpu_indexes)]
    subordinates: list = field(default_factory=list)
    # CPU weight offload (reference: GGUF partial offload / offload_layers
    # in the claim, schemas/models.py:623-630): host GiB streamed per step
    ram_claim: int = 2 << 30
    offload_gb: float = 0.0
    offload_layers: int = 0


def model_spec_for(model: Model | dict) -> ModelSpec | None:
    source = model["source"] if isinstance(model, dict) else model.source
    ref = model["model_ref"] if isinstance(model, dict) else model.model_ref
    if source == "preset":
        return PRESETS.get(ref)
    try:
        if str(ref).endswith(".gguf"):
            from ..utils.gguf import spec_from_gguf

            return spec_from_gguf(ref)
        import json as _json
        from pathlib import Path as _Path

        with open(_Path(ref) / "config.json") as f:
            arch = (_json.load(f).get("architectures") or [""])[0]
        from ..models.encoder import is_encoder_arch

        if is_encoder_arch(arch):
            # cross-encoder reranker: synthesize an LLM-shaped spec whose
            # weight_bytes matches the encoder (placement sizing only —
            # the engine server builds the real EncoderSpec)
            from ..models.encoder import EncoderSpec

            e = EncoderSpec.from_dir(ref)
            return ModelSpec(
                architecture=e.architecture, vocab_size=e.vocab_size,
                hidden_size=e.hidden_size,
                intermediate_size=e.intermediate_size,
                num_layers=e.num_layers, num_heads=e.num_heads,
                num_kv_heads=e.num_heads,
                head_dim=e.hidden_size // e.num_heads,
                max_position_embeddings=e.max_position_embeddings,
                tie_word_embeddings=True,
            )
        return ModelSpec.from_dir(ref)
    except Exception:  # noqa: BLE001
        return None


def _weight_bytes(model: dict, spec: ModelSpec, tp: int) -> int:
    """Per-shard resident weight bytes, accounting for runtime W4 packing
    (backend_parameters quantize_runtime=w4 keeps weights int4: ~0.28x of
    bf16 incl. scales) and the bf16 default."""
    bp = model.get("backend_parameters") or {}
    w = spec.weight_bytes() // tp
    if bp.get("quantize_runtime") == "w4":
        w = int(w * 0.28)
    return w


def estimate_vram_claim(model: Model | dict, spec: ModelSpec | None, tp: int) -> int:
    """Minimum per-GPU VRAM need (bytes) for one replica shard: exact
    weights + framework/activation overhead + KV-pool floor. The FULL
    claim recorded on placement additionally includes the KV pool the
    engine will actually take (gmu x remaining free) — see
    claim_for_allocatable (reference memory model:
    policies/utils.py:384-470 + vllm_resource_fit_selector.py:166-206)."""
    if isinstance(model, Model):
        model = model.to_dict()
    if spec is None:
        return 16 << 30
    weights = int(_weight_bytes(model, spec, tp) * WEIGHT_FUDGE) + FRAMEWORK_OVERHEAD
    return weights + MIN_KV_BYTES


def claim_for_allocatable(model: dict, spec: ModelSpec | None, tp: int,
                          alloc_bytes: int) -> int:
    """What the engine will actually consume on a GPU with `alloc_bytes`
    free: weights + overhead + gmu x (free - weights - overhead). Claiming
    only the floor would let a later placement over-commit the device the
    moment this engine sizes its KV pool."""
    need = estimate_vram_claim(model, spec, tp)
    if spec is None:
        return need
    gmu = model.get("gpu_memory_utilization") or 0.9
    base = need - MIN_KV_BYTES
    kv = max(MIN_KV_BYTES, int((alloc_bytes - base) * gmu))
    return min(alloc_bytes, base + kv)


# ---- filters (reference: policies/worker_filters/*) -----------------------

def status_filter(workers: list[dict], model: dict) -> list[dict]:
    return [w for w in workers if w.get("state") == WorkerState.READY.value]


def label_filter(workers: list[dict], model: dict) -> list[dict]:
    sel = model.get("worker_selector") or {}
    if not sel:
        return workers
    return [
        w for w in workers
        if all((w.get("labels") or {}).get(k) == v for k, v in sel.items())
    ]


def gpu_arch_filter(workers: list[dict], model: dict) -> list[dict]:
    """MI355X-native engine runs on gfx950 (rocm) devices."""
    out = []
    for w in workers:
        devs = (w.get("status") or {}).get("gpu_devices", [])
        if any(d.get("type") == "rocm" for d in devs):
            out.append(w)
    return out


def cluster_filter(workers: list[dict], model: dict) -> list[dict]:
    """Multi-cluster scoping (reference: policies/worker_filters
    ClusterFilter): a model pinned to a cluster only places on that
    cluster's workers; unpinned models see every worker."""
    cid = model.get("cluster_id")
    if not cid:
        return workers
    return [w for w in workers if w.get("cluster_id") == cid]


FILTER_CHAIN = [cluster_filter, status_filter, label_filter, gpu_arch_filter]


# ---- allocation accounting ------------------------------------------------

def worker_allocatable(worker: dict, instances: list[dict]) -> dict[int, int]:
    """Free VRAM per GPU index after subtracting recorded claims
    (claim-based, not live — reference policies/utils.py:91-150)."""
    reserved = (worker.get("system_reserved") or {}).get("vram", 0)
    out: dict[int, int] = {}
    for d in (worker.get("status") or {}).get("gpu_devices", []):
        total = (d.get("memory") or {}).get("total", 0)
        out[d.get("index", 0)] = max(0, total - reserved)
    for inst in instances:
        if inst.get("worker_id") != worker.get("id"):
            continue
        claim = (inst.get("computed_resource_claim") or {}).get("vram", {})
        for idx_str, bytes_ in claim.items():
            idx = int(idx_str)
            if idx in out:
                out[idx] = max(0, out[idx] - bytes_)
    return out


# ---- selector -------------------------------------------------------------

def device_matches_type(dev: dict, sel: dict | None) -> bool:
    """Device-class constraint (reference gpu_type_selector — vGPU slice /
    MIG partition classes, schemas/models.py:92-175; here the MI355X-native
    classes: AMD compute-partition mode SPX/DPX/CPX, NPS memory mode,
    minimum VRAM, device-name substring)."""
    if not sel:
        return True
    part = dev.get("partition") or {}
    want_c = sel.get("partition_compute")
    if want_c and (part.get("compute") or "").upper() != want_c.upper():
        return False
    want_m = sel.get("partition_memory")
    if want_m and (part.get("memory") or "").upper() != want_m.upper():
        return False
    min_gb = sel.get("min_vram_gb")
    if min_gb and ((dev.get("memory") or {}).get("total", 0)
                   < min_gb * (1 << 30)):
        return False
    sub = sel.get("name_contains")
    if sub and sub.lower() not in (dev.get("name") or "").lower():
        return False
    return True


def select_candidates(model: dict, workers: list[dict], instances: list[dict]) -> list[Candidate]:
    spec = model_spec_for(model)
    tp = max(1, model.get("gpus_per_replica") or 1)
    claim = estimate_vram_claim(model, spec, tp)   # per-GPU floor
    manual = model.get("gpu_selector") or None
    type_sel = model.get("gpu_type_selector") or None

    def full_claim(alloc: dict[int, int], picks: list[int]) -> dict[int, int]:
        return {i: claim_for_allocatable(model, spec, tp, alloc[i])
                for i in picks}

    out: list[Candidate] = []
    for w in workers:
        alloc = worker_allocatable(w, instances)
        if type_sel:
            devs = {d.get("index", 0): d
                    for d in (w.get("status") or {}).get("gpu_devices", [])}
            alloc = {i: free for i, free in alloc.items()
                     if device_matches_type(devs.get(i, {}), type_sel)}
        if manual:
            ids = manual.get("gpu_ids", [])
            picks = []
            for gid in ids:
                parts = str(gid).split(":")
                if parts[0] in (str(w.get("id")), w.get("name")):
                    picks.append(int(parts[-1]))
            if len(picks) >= tp and all(alloc.get(i, 0) >= claim for i in picks[:tp]):
                out.append(Candidate(w, picks[:tp],
                                     vram_claim=full_claim(alloc, picks[:tp])))
            continue
        fits = sorted(
            (i for i, free in alloc.items() if free >= claim),
            key=lambda i: -alloc[i],
        )
        if len(fits) >= tp:
            picks = fits[:tp]
            out.append(Candidate(w, picks, vram_claim=full_claim(alloc, picks)))
    if not out and model.get("distributed_inference_across_workers"):
        cand = _multi_worker_candidate(tp, claim, workers, instances)
        if cand is not None:
            out.append(cand)
    if not out and tp == 1:
        cand = _offload_candidate(model, spec, workers, instances)
        if cand is not None:
            out.append(cand)
    return out


def _offload_candidate(model: dict, spec: ModelSpec | None,
                       workers: list[dict],
                       instances: list[dict]) -> Candidate | None:
    """Partial CPU offload placement (reference: the GGUF selector's
    layer-offload path, gguf_resource_fit_selector.py:129-300 +
    offload_layer_scorer): when no GPU holds the full weights, place on
    the roomiest GPU and stream the overflow layers' weights from pinned
    host DRAM (engine/offload.py). Opt-in via backend_parameters
    cpu_offload=true (matches the reference's explicit offload knobs)."""
    bp = model.get("backend_parameters") or {}
    if not bp.get("cpu_offload") or spec is None:
        return None
    weights = int(_weight_bytes(model, spec, 1) * WEIGHT_FUDGE)
    need_floor = FRAMEWORK_OVERHEAD + MIN_KV_BYTES
    best: tuple[int, dict, int] | None = None
    for w in workers:
        alloc = worker_allocatable(w, instances)
        for i, free in alloc.items():
            if free > need_floor and (best is None or free > best[0]):
                best = (free, w, i)
    if best is None:
        return None
    free, w, idx = best
    resident_budget = int(free * 0.9) - need_floor
    if resident_budget <= 0 or weights <= resident_budget:
        return None
    offload_bytes = weights - resident_budget
    if offload_bytes > weights * 0.9:
        return None  # less than 10% resident: refuse (would crawl)
    per_layer = weights / max(1, spec.num_layers)
    layers = min(spec.num_layers - 1,
                 max(1, int(offload_bytes / per_layer + 0.999)))
    host_ram = (w.get("status") or {}).get("memory", {}).get("total", 0)
    ram_claim = offload_bytes + (2 << 30)
    if host_ram and ram_claim > host_ram:
        return None
    cand = Candidate(w, [idx], vram_claim={idx: int(free * 0.9)})
    cand.offload_gb = round(offload_bytes / 2**30 + 0.05, 2)
    cand.offload_layers = layers
    cand.ram_claim = ram_claim
    return cand


def _multi_worker_candidate(tp: int, claim: int, workers: list[dict],
                            instances: list[dict]) -> Candidate | None:
    """Equal-GPU-count worker groups (reference selector semantics): find
    the smallest worker count nw where tp/nw GPUs fit on each of nw
    workers; first worker is the main (rank 0), the rest subordinate."""
    fits_per_worker = []
    for w in workers:
        alloc = worker_allocatable(w, instances)
        fits = sorted((i for i, free in alloc.items() if free >= claim),
                      key=lambda i: -alloc[i])
        if fits:
            fits_per_worker.append((w, fits))
    for nw in range(2, len(fits_per_worker) + 1):
        if tp % nw:
            continue
        per = tp // nw
        group = [(w, f[:per]) for w, f in fits_per_worker if len(f) >= per]
        if len(group) >= nw:
            group = group[:nw]
            main_w, main_g = group[0]
            cand = Candidate(main_w, main_g,
                             vram_claim={i: claim for i in main_g})
            cand.subordinates = group[1:]
            return cand
    return None


# ---- scorers (reference: policies/scorers/*) ------------------------------

def placement_score(cand: Candidate, model: dict, instances: list[dict]) -> float:
    """binpack: prefer busier workers; spread: prefer emptier ones."""
    strategy = model.get("placement_strategy", PlacementStrategy.BINPACK.value)
    alloc = worker_allocatable(cand.worker, instances)
    total = sum(
        (d.get("memory") or {}).get("total", 1)
        for d in (cand.worker.get("status") or {}).get("gpu_devices", [])
    ) or 1
    free_frac = sum(alloc.values()) / total
    if strategy == PlacementStrategy.SPREAD.value:
        return free_frac * 100
    return (1 - free_frac) * 100


def replica_spread_score(cand: Candidate, model: dict, instances: list[dict]) -> float:
    """Prefer workers with fewer replicas of the same model (spreads load
    across the node set at equal placement scores)."""
    n = sum(
        1 for i in instances
        if i.get("model_id") == model.get("id") and i.get("worker_id") == cand.worker.get("id")
    )
    return -n


def file_locality_score(cand: Candidate, model: dict,
                        model_files: list[dict] | None) -> float:
    """Prefer workers that already hold the model's files (reference:
    policies/scorers/model_file_locality_scorer.py) — skips a multi-GB
    download before STARTING."""
    if not model_files:
        return 0.0
    wid = cand.worker.get("id")
    for f in model_files:
        if (f.get("worker_id") == wid
                and f.get("model_ref") == model.get("model_ref")
                and f.get("source") == model.get("source")):
            return 10.0
    return 0.0


def pick_candidate(model: dict, workers: list[dict], instances: list[dict],
                   model_files: list[dict] | None = None) -> Candidate | None:
    flt = workers
    for f in FILTER_CHAIN:
        flt = f(flt, model)
        if not flt:
            return None
    cands = select_candidates(model, flt, instances)
    if not cands:
        return None
    for c in cands:
        c.score = (placement_score(c, model, instances)
                   + replica_spread_score(c, model, instances)
                   + file_locality_score(c, model, model_files))
    return max(cands, key=lambda c: c.score)
