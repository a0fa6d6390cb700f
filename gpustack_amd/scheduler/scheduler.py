"""Placement scheduler loop (reference: gpustack/scheduler/scheduler.py:85).

Event-driven (instance CREATED) + periodic scan of PENDING instances; for
each: ANALYZING (resolve ModelSpec / claims) -> find candidate via the
policy chain -> write worker_id / gpu_indexes / claim -> SCHEDULED.
Stale SCHEDULED instances (worker never picked them up) are retried
(reference: scheduler.py:262-299)."""
from __future__ import annotations

import logging
import queue
import time

from ..config import Config
from ..db import EventType, ar_update, bus, get_session
from ..schemas import Model, ModelInstance, ModelInstanceState, Worker
from .policies import estimate_vram_claim, model_spec_for, pick_candidate

logger = logging.getLogger(__name__)

STALE_SCHEDULED_SECONDS = 180.0


class PlacementScheduler:
    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._stop = False
        self.coordinator = None  # set by the server; None = always leader

    def _is_leader(self) -> bool:
        return self.coordinator is None or self.coordinator.is_leader

    def stop(self):
        self._stop = True

    def run(self) -> None:
        q = bus.subscribe("model_instances")
        last_scan = 0.0
        while not self._stop:
            try:
                try:
                    ev = q.get(timeout=5.0)
                    if ev.type == EventType.CREATED and self._is_leader():
                        self.schedule_one(ev.data["id"])
                except queue.Empty:
                    pass
                if not self._is_leader():
                    continue
                if time.time() - last_scan > 15.0:
                    last_scan = time.time()
                    self.scan()
            except Exception:  # noqa: BLE001
                logger.exception("scheduler cycle failed")
                time.sleep(2)

    def scan(self) -> None:
        now = time.time()
        with get_session() as s:
            pending = [
                i.id for i in s.query(ModelInstance)
                .filter(ModelInstance.state.in_([
                    ModelInstanceState.PENDING.value,
                    ModelInstanceState.ANALYZING.value,
                ])).all()
            ]
            stale = [
                i.id for i in s.query(ModelInstance)
                .filter_by(state=ModelInstanceState.SCHEDULED.value).all()
                if now - i.updated_at > STALE_SCHEDULED_SECONDS
            ]
        for iid in pending + stale:
            self.schedule_one(iid)

    @staticmethod
    def _pp_partition(model_d, spec, cand, workers, others):
        """Per-stage layer counts proportional to each stage's free VRAM
        when backend_parameters requests pipeline parallelism over GPUs
        with unequal headroom. None for even/trivial cases (the engine's
        default even split applies)."""
        bp = model_d.get("backend_parameters") or {}
        try:
            pp = int(bp.get("pp_size", 1) or 1)
        except (TypeError, ValueError):
            return None
        gpus = cand.gpu_indexes
        if (pp <= 1 or spec is None or cand.subordinates
                or not gpus or len(gpus) % pp):
            return None
        from .policies import worker_allocatable

        alloc = worker_allocatable(cand.worker, others)
        tp = len(gpus) // pp
        stage_free = [sum(alloc.get(g, 0) for g in gpus[s * tp:(s + 1) * tp])
                      for s in range(pp)]
        total = sum(stage_free)
        if total <= 0:
            return None
        n = spec.num_layers
        part = [max(1, int(n * f / total)) for f in stage_free]
        # fix rounding: add/remove from the roomiest/most-loaded stages
        while sum(part) < n:
            part[stage_free.index(max(stage_free))] += 1
        while sum(part) > n:
            i = max(range(pp), key=lambda j: part[j])
            if part[i] <= 1:
                return None
            part[i] -= 1
        if part == [n // pp + (1 if i < n % pp else 0) for i in range(pp)]:
            return None  # even: engine default
        return part


    def schedule_one(self, instance_id: int) -> bool:
        with get_session() as s:
            inst = s.get(ModelInstance, instance_id)
            if inst is None or inst.state not in (
                ModelInstanceState.PENDING.value,
                ModelInstanceState.ANALYZING.value,
                ModelInstanceState.SCHEDULED.value,
            ):
                return False
            model = s.get(Model, inst.model_id)
            if model is None:
                return False
            model_d = model.to_dict()
            spec = model_spec_for(model_d)
            if spec is None:
                inst.state = ModelInstanceState.ANALYZING.value
                inst.state_message = f"cannot resolve model spec for {model.model_ref!r}"
                ar_update(s, inst)
                return False
            workers = [w.to_dict() for w in s.query(Worker).all()]
            others = [
                i.to_dict() for i in s.query(ModelInstance).all() if i.id != inst.id
            ]
            from ..schemas import ModelFile

            files = [f.to_dict() for f in s.query(ModelFile).all()]
            cand = pick_candidate(model_d, workers, others, files)
            if cand is None:
                inst.state = ModelInstanceState.ANALYZING.value
                inst.state_message = "no worker fits the resource claim"
                ar_update(s, inst)
                return False
            tp = max(1, model.gpus_per_replica or 1)
            fallback = estimate_vram_claim(model_d, spec, tp)
            inst.worker_id = cand.worker["id"]
            inst.worker_ip = cand.worker.get("ip", "")
            inst.gpu_indexes = cand.gpu_indexes
            inst.computed_resource_claim = {
                "vram": {str(i): cand.vram_claim.get(i, fallback)
                         for i in cand.gpu_indexes},
                "ram": cand.ram_claim,
            }
            if cand.offload_gb > 0:
                # partial CPU offload (reference claim fields:
                # schemas/models.py:623-630 offload_layers)
                inst.computed_resource_claim["offload_gb"] = cand.offload_gb
                inst.computed_resource_claim["offload_layers"] = cand.offload_layers
            part = self._pp_partition(model_d, spec, cand, workers, others)
            if part:
                # uneven pipeline stages sized to each GPU group's free
                # VRAM — the native analog of the reference's per-GPU GGUF
                # tensor_split proportions (gguf selector :370-441)
                inst.computed_resource_claim["pp_partition"] = part
            if cand.subordinates:
                # cross-worker TP: rank layout + rendezvous port
                # (reference: serve_manager.py:1643-1739 port bands +
                # distributed_servers subordinate list)
                ranks = []
                base = len(cand.gpu_indexes)
                for w, gpus in cand.subordinates:
                    ranks.append({
                        "worker_id": w["id"],
                        "worker_ip": w.get("ip", ""),
                        "gpu_indexes": gpus,
                        "rank_base": base,
                    })
                    base += len(gpus)
                # rendezvous port from a dedicated band, skipping ports any
                # live instance already holds (reference: distributed port
                # bands, serve_manager.py:1643-1739 — chosen here because
                # subordinate workers must agree on it before launch)
                taken = {
                    (i.get("distributed_servers") or {}).get("master_port")
                    for i in others
                }
                mp = 45000 + (inst.id * 4) % 1000
                for _ in range(250):  # band holds 250 slots of 4
                    if mp not in taken:
                        break
                    mp = 45000 + (mp - 45000 + 4) % 1000
                inst.distributed_servers = {
                    "tp": tp,
                    "master_ip": cand.worker.get("ip", ""),
                    "master_port": mp,
                    "subordinates": ranks,
                }
            inst.state = ModelInstanceState.SCHEDULED.value
            inst.state_message = ""
            ar_update(s, inst)
            logger.info(
                "scheduled %s -> worker %s gpus %s",
                inst.name, cand.worker.get("name"), cand.gpu_indexes,
            )
            return True
