"""Reconcilers (reference: gpustack/server/controllers.py).

ModelController.sync_replicas (controllers.py:300): creates/deletes
ModelInstances to match Model.replicas.
WorkerMonitor (worker_syncer.py:51-134 + controllers.py:1399): flips stale
workers NOT_READY/UNREACHABLE and marks their instances UNREACHABLE.
SystemLoadCollector (system_load.py:113): periodic cluster utilization
snapshots.
"""
from __future__ import annotations

import logging
import queue
import time

from ..config import Config
from ..db import EventType, ar_create, ar_delete, ar_update, bus, get_session
from ..schemas import (
    Model, ModelInstance, ModelInstanceState, SystemLoad, Worker, WorkerState,
)

logger = logging.getLogger(__name__)

HEARTBEAT_TIMEOUT = 60.0
UNREACHABLE_TIMEOUT = 180.0


class _LeaderGated:
    coordinator = None  # set by the server; None = always leader

    def _is_leader(self) -> bool:
        return self.coordinator is None or self.coordinator.is_leader


def model_spec_hash(model) -> str:
    """Hash of every field that changes how an instance must be served;
    instances carrying a stale hash are replaced by the controller."""
    import hashlib
    import json as _json

    payload = _json.dumps({
        "source": model.source, "model_ref": model.model_ref,
        "gpus_per_replica": model.gpus_per_replica,
        "backend_parameters": model.backend_parameters,
        "env": model.env, "max_model_len": model.max_model_len,
        "gpu_memory_utilization": model.gpu_memory_utilization,
        "speculative_config": model.speculative_config,
        "extended_kv_cache": model.extended_kv_cache,
        "lora_list": model.lora_list,
        "lora_adapters": model.lora_adapters,
        "distributed": model.distributed_inference_across_workers,
    }, sort_keys=True, default=str)
    return hashlib.sha256(payload.encode()).hexdigest()[:16]


class ModelController(_LeaderGated):
    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def run(self) -> None:
        from .workqueue import WorkQueue

        q = bus.subscribe("models")
        iq = bus.subscribe("model_instances")
        # rate-limited queue (reference: server/workqueue.py): coalesces
        # bursts of events per model and retries failing reconciles with
        # exponential backoff instead of hot-looping
        self.wq = WorkQueue(base_delay=2.0, max_delay=300.0)
        self.reconcile_all()
        while not self._stop:
            try:
                ev = q.get(timeout=0.5)
                if self._stop:
                    return
                if ev.type in (EventType.CREATED, EventType.UPDATED) and self._is_leader():
                    self.wq.add(ev.data["id"])
            except queue.Empty:
                pass
            # drain instance deletions (e.g. user deleted an instance -> recreate)
            try:
                while True:
                    iev = iq.get_nowait()
                    if self._stop:
                        return
                    if iev.type == EventType.DELETED and iev.data.get("model_id"):
                        self.wq.add(iev.data["model_id"])
            except queue.Empty:
                pass
            item = self.wq.get(timeout=0.01)
            if item is None:
                continue
            try:
                self.sync_replicas(item)
                self.wq.done(item)
            except Exception:  # noqa: BLE001
                logger.exception("sync_replicas(%s) failed; backing off", item)
                self.wq.done(item, requeue=True)

    def reconcile_all(self) -> None:
        with get_session() as s:
            ids = [m.id for m in s.query(Model).all()]
        for mid in ids:
            try:
                self.sync_replicas(mid)
            except Exception:  # noqa: BLE001
                logger.exception("sync_replicas(%s) failed", mid)

    def sync_replicas(self, model_id: int) -> None:
        self.sync_replicas_to(model_id, None)

    def sync_replicas_to(self, model_id: int, want_override: int | None) -> None:
        with get_session() as s:
            model = s.get(Model, model_id)
            if model is None:
                return
            insts = s.query(ModelInstance).filter_by(model_id=model_id).all()
            want = model.replicas if want_override is None else want_override
            # model updates redeploy: drop instances built from a stale spec
            # (their deletion events re-enter this loop and recreate them)
            cur_hash = model_spec_hash(model)
            stale = [i for i in insts if i.spec_hash and i.spec_hash != cur_hash]
            for v in stale:
                ar_delete(s, v)
            insts = [i for i in insts if i not in stale]
            have = len(insts)
            if have < want:
                used = {i.name for i in insts}
                for n in range(want * 2):
                    if have >= want:
                        break
                    name = f"{model.name}-{n}"
                    if name in used:
                        continue
                    inst = ModelInstance(
                        model_id=model.id, model_name=model.name, name=name,
                        state=ModelInstanceState.PENDING.value,
                        spec_hash=cur_hash,
                    )
                    ar_create(s, inst)
                    have += 1
            elif have > want:
                # scale down: prefer non-RUNNING victims (StatusScorer semantics)
                order = {st.value: i for i, st in enumerate([
                    ModelInstanceState.ERROR, ModelInstanceState.UNREACHABLE,
                    ModelInstanceState.PENDING, ModelInstanceState.ANALYZING,
                    ModelInstanceState.SCHEDULED, ModelInstanceState.INITIALIZING,
                    ModelInstanceState.DOWNLOADING, ModelInstanceState.STARTING,
                    ModelInstanceState.RUNNING,
                ])}
                victims = sorted(insts, key=lambda i: order.get(i.state, 9))
                for v in victims[: have - want]:
                    ar_delete(s, v)


class WorkerMonitor(_LeaderGated):
    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def run(self) -> None:
        while not self._stop:
            try:
                if self._is_leader():
                    self.check_once()
            except Exception:  # noqa: BLE001
                logger.exception("worker monitor cycle failed")
            for _ in range(15):
                if self._stop:
                    return
                time.sleep(1.0)

    def check_once(self) -> None:
        now = time.time()
        with get_session() as s:
            for w in s.query(Worker).all():
                age = now - (w.heartbeat_time or 0)
                if age > UNREACHABLE_TIMEOUT and w.state != WorkerState.UNREACHABLE.value:
                    w.state = WorkerState.UNREACHABLE.value
                    w.state_message = f"no heartbeat for {int(age)}s"
                    ar_update(s, w)
                    self._mark_instances(s, w.id)
                elif HEARTBEAT_TIMEOUT < age <= UNREACHABLE_TIMEOUT and w.state == WorkerState.READY.value:
                    w.state = WorkerState.NOT_READY.value
                    w.state_message = f"no heartbeat for {int(age)}s"
                    ar_update(s, w)

    def _mark_instances(self, s, worker_id: int) -> None:
        for inst in s.query(ModelInstance).filter_by(worker_id=worker_id).all():
            if inst.state == ModelInstanceState.RUNNING.value:
                inst.state = ModelInstanceState.UNREACHABLE.value
                inst.state_message = "worker unreachable"
                ar_update(s, inst)


class ScalingScheduler(_LeaderGated):
    """Cron-window desired-replica computation (reference:
    server/scaling_scheduler.py:19,96). While a window is active the
    model's instance count follows the rule's replicas; outside windows
    it returns to the baseline Model.replicas."""

    def __init__(self, cfg: Config, interval: float = 30.0):
        self.cfg = cfg
        self.interval = interval
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def desired_replicas(self, model) -> int:
        from ..utils.cron import window_active

        sched = model.scaling_schedule or {}
        desired = model.replicas
        for rule in sched.get("rules", []):
            try:
                if window_active(rule["cron"], int(rule.get("duration_minutes", 60))):
                    desired = max(desired, int(rule.get("replicas", desired)))
            except (KeyError, ValueError):
                logger.warning("bad scaling rule on model %s: %s", model.name, rule)
        return desired

    def run(self) -> None:
        mc = ModelController(self.cfg)
        while not self._stop:
            try:
                if not self._is_leader():
                    raise StopIteration  # skip cycle, keep sleeping
                with get_session() as s:
                    models = [m for m in s.query(Model).all() if m.scaling_schedule]
                for m in models:
                    want = self.desired_replicas(m)
                    with get_session() as s:
                        have = s.query(ModelInstance).filter_by(model_id=m.id).count()
                    if have != want:
                        mc.sync_replicas_to(m.id, want)
            except StopIteration:
                pass
            except Exception:  # noqa: BLE001
                logger.exception("scaling scheduler cycle failed")
            for _ in range(int(self.interval)):
                if self._stop:
                    return
                time.sleep(1.0)


class ResourceEventLogger(_LeaderGated):
    """Metering event writer (reference: ResourceEventLogger wired at
    server/server.py:541-595): subscribes to ModelInstance bus events and
    appends one resource_events row per STATE TRANSITION with the claim
    footprint attached — the hot half of the hot+archive pair the
    UsageArchiver drains."""

    METERED_STATES = {"scheduled", "running", "stopped", "error",
                      "unreachable"}

    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._stop = False
        self._last_state: dict[int, str] = {}

    def stop(self) -> None:
        self._stop = True

    def _record(self, data: dict) -> None:
        from ..schemas.tables import ResourceEvent

        iid = data.get("id")
        state = data.get("state")
        if iid is None or state not in self.METERED_STATES:
            return
        if self._last_state.get(iid) == state:
            return  # non-transition update (heartbeat field churn)
        self._last_state[iid] = state
        claim = data.get("computed_resource_claim") or {}
        vram = sum((claim.get("vram") or {}).values())
        with get_session() as s:
            s.add(ResourceEvent(
                event_type=state, instance_id=iid,
                model_id=data.get("model_id"),
                model_name=data.get("model_name", ""),
                worker_id=data.get("worker_id"),
                gpu_indexes=data.get("gpu_indexes") or [],
                vram_bytes=int(vram), ram_bytes=int(claim.get("ram") or 0),
            ))
            s.commit()

    def run(self) -> None:
        import queue as _q

        from ..db import bus

        q = bus.subscribe("model_instances")
        try:
            while not self._stop:
                try:
                    ev = q.get(timeout=1.0)
                except _q.Empty:
                    continue
                try:
                    if self._is_leader() and isinstance(ev.data, dict):
                        if ev.type.value == "DELETED":
                            self._last_state.pop(ev.data.get("id"), None)
                        else:
                            self._record(ev.data)
                except Exception:  # noqa: BLE001
                    logger.exception("resource-event record failed")
        finally:
            bus.unsubscribe("model_instances", q)


class UsageArchiver(_LeaderGated):
    """Hot -> archive mover for usage rows older than `keep_days`
    (reference: server/usage_archiver.py TableArchiver)."""

    def __init__(self, cfg: Config, keep_days: int = 30, interval: float = 3600.0):
        self.cfg = cfg
        self.keep_days = keep_days
        self.interval = interval
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def archive_once(self) -> int:
        import datetime as dt

        from ..schemas import ModelUsage
        from ..schemas.tables import ModelUsageArchive

        cutoff = (dt.date.today() - dt.timedelta(days=self.keep_days)).isoformat()
        moved = 0
        with get_session() as s:
            for row in s.query(ModelUsage).filter(ModelUsage.date < cutoff).all():
                s.add(ModelUsageArchive(
                    user_id=row.user_id, model_id=row.model_id,
                    model_name=row.model_name, date=row.date,
                    prompt_tokens=row.prompt_tokens,
                    completion_tokens=row.completion_tokens,
                    request_count=row.request_count,
                ))
                s.delete(row)
                moved += 1
            # resource-event pair (same retention window, timestamp-based)
            import time as _t

            from ..schemas.tables import ResourceEvent, ResourceEventArchive

            ts_cutoff = _t.time() - self.keep_days * 86400
            for row in s.query(ResourceEvent).filter(
                    ResourceEvent.timestamp < ts_cutoff).all():
                s.add(ResourceEventArchive(
                    event_type=row.event_type, instance_id=row.instance_id,
                    model_id=row.model_id, model_name=row.model_name,
                    worker_id=row.worker_id, gpu_indexes=row.gpu_indexes,
                    vram_bytes=row.vram_bytes, ram_bytes=row.ram_bytes,
                    timestamp=row.timestamp,
                ))
                s.delete(row)
                moved += 1
            s.commit()
        return moved

    def run(self) -> None:
        while not self._stop:
            try:
                n = self.archive_once() if self._is_leader() else 0
                if n:
                    logger.info("archived %d usage rows", n)
            except Exception:  # noqa: BLE001
                logger.exception("usage archiver cycle failed")
            for _ in range(int(self.interval)):
                if self._stop:
                    return
                time.sleep(1.0)


class SystemLoadCollector(_LeaderGated):
    def __init__(self, cfg: Config, interval: float = 60.0):
        self.cfg = cfg
        self.interval = interval
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def run(self) -> None:
        while not self._stop:
            try:
                if self._is_leader():
                    self.collect_once()
            except Exception:  # noqa: BLE001
                logger.exception("system load collection failed")
            for _ in range(int(self.interval)):
                if self._stop:
                    return
                time.sleep(1.0)

    def collect_once(self) -> None:
        with get_session() as s:
            workers = s.query(Worker).filter_by(state=WorkerState.READY.value).all()
            if not workers:
                return
            cpu = ram = gpu = vram = 0.0
            n_gpu = 0
            for w in workers:
                st = w.status or {}
                cpu += st.get("cpu", {}).get("utilization_rate", 0.0)
                mem = st.get("memory", {})
                if mem.get("total"):
                    ram += mem.get("used", 0) / mem["total"] * 100
                for d in st.get("gpu_devices", []):
                    n_gpu += 1
                    gpu += d.get("core", {}).get("utilization_rate", 0.0)
                    dm = d.get("memory", {})
                    if dm.get("total"):
                        vram += dm.get("used", 0) / dm["total"] * 100
            nw = len(workers)
            s.add(SystemLoad(
                cpu=cpu / nw, ram=ram / nw,
                gpu=gpu / n_gpu if n_gpu else 0.0,
                vram=vram / n_gpu if n_gpu else 0.0,
            ))
            s.commit()


class WorkerPoolController(_LeaderGated):
    """Reconciles WorkerPool.replicas against provisioned instances
    (reference: WorkerPoolController + WorkerProvisioningController,
    server/controllers.py:2352,2398 — cloud scale-out). Providers live in
    server/providers.py; the bootstrap script is the cloud-init analog."""

    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def run(self) -> None:
        from ..schemas import WorkerPool  # noqa: F401

        q = bus.subscribe("worker_pools")
        self.reconcile_all()
        while not self._stop:
            try:
                ev = q.get(timeout=30.0)
                if self._stop:
                    return
                if ev.type in (EventType.CREATED, EventType.UPDATED) and self._is_leader():
                    self.reconcile_pool(ev.data["id"])
            except queue.Empty:
                if self._is_leader():
                    self.reconcile_all()

    def reconcile_all(self) -> None:
        from ..schemas import WorkerPool

        with get_session() as s:
            ids = [p.id for p in s.query(WorkerPool).all()]
        for pid in ids:
            try:
                self.reconcile_pool(pid)
            except Exception:  # noqa: BLE001
                logger.exception("worker pool %s reconcile failed", pid)

    def _bootstrap(self, pool) -> str:
        from ..schemas import RegistrationToken
        from .providers import bootstrap_script

        with get_session() as s:
            row = s.query(RegistrationToken).first()
            token = row.token if row else ""
        server_url = f"http://{self.cfg.host}:{self.cfg.port}"
        return bootstrap_script(server_url, token, pool.labels)

    def reconcile_pool(self, pool_id: int) -> None:
        from ..schemas import WorkerPool
        from .providers import get_provider

        with get_session() as s:
            pool = s.get(WorkerPool, pool_id)
            if pool is None:
                return
            try:
                provider = get_provider(pool.provider, pool.provider_config)
            except ValueError as e:
                pool.state_message = str(e)
                ar_update(s, pool)
                return
            # copy the dicts too: mutating the loaded JSON in place would
            # erase the attribute history and skip the flush
            records = [dict(r) for r in (pool.instances or [])]
            want = pool.replicas or 0
            changed = False
            errors: list[str] = []
            while len(records) < want:
                name = f"{pool.name}-{len(records)}"
                taken = {r["name"] for r in records}
                n = 0
                while name in taken:
                    n += 1
                    name = f"{pool.name}-{n + len(records)}"
                try:
                    iid = provider.create(name, pool.instance_type,
                                          self._bootstrap(pool))
                except Exception as e:  # noqa: BLE001
                    errors.append(f"create failed: {e}")
                    break
                records.append({"instance_id": iid, "name": name,
                                "state": "provisioning",
                                "created_at": time.time()})
                changed = True
            while len(records) > want:
                # scale down newest-first; workers that registered from the
                # node disappear via the heartbeat monitor once it powers off
                victim = records[-1]
                try:
                    provider.delete(victim["instance_id"])
                except Exception as e:  # noqa: BLE001
                    errors.append(f"delete failed: {e}")
                    break
                records.pop()
                changed = True
            # mark provisioned instances whose worker has registered READY
            with_workers = {w.name for w in s.query(Worker).all()}
            for r in records:
                new_state = "ready" if r["name"] in with_workers else r["state"]
                if new_state != r["state"]:
                    r["state"] = new_state
                    changed = True
            if changed or errors:
                pool.instances = records
                pool.state_message = "; ".join(errors)
                ar_update(s, pool)
