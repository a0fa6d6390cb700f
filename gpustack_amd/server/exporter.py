"""Server-side Prometheus exporter (reference: gpustack/exporter/exporter.py).

Wire-compatible metric family names with the reference where the concept
maps 1:1 (gpustack_worker_*, gpustack_model_*); cluster aggregates come
from the DB like the reference's DB-driven exporter."""
from __future__ import annotations

from prometheus_client import CollectorRegistry, Gauge, generate_latest

from ..db import get_session
from ..schemas import Model, ModelInstance, ModelInstanceState, ModelUsage, Worker, WorkerState


def render_metrics() -> bytes:
    reg = CollectorRegistry()
    g_workers = Gauge("gpustack_workers", "workers by state", ["state"], registry=reg)
    g_gpu_util = Gauge("gpustack_worker_gpu_utilization_rate", "GPU core utilization",
                       ["worker", "index"], registry=reg)
    g_vram_total = Gauge("gpustack_worker_gpu_vram_total_bytes", "GPU VRAM total",
                         ["worker", "index"], registry=reg)
    g_vram_used = Gauge("gpustack_worker_gpu_vram_used_bytes", "GPU VRAM used",
                        ["worker", "index"], registry=reg)
    g_vram_alloc = Gauge("gpustack_worker_gpu_vram_allocated_bytes", "GPU VRAM allocated (claims)",
                         ["worker", "index"], registry=reg)
    g_instances = Gauge("gpustack_model_instances", "instances by state",
                        ["model", "state"], registry=reg)
    g_replicas = Gauge("gpustack_model_desired_replicas", "desired replicas",
                       ["model"], registry=reg)
    g_prompt = Gauge("gpustack_model_prompt_tokens_total", "prompt tokens",
                     ["model"], registry=reg)
    g_completion = Gauge("gpustack_model_completion_tokens_total", "completion tokens",
                         ["model"], registry=reg)
    g_requests = Gauge("gpustack_model_requests_total", "requests", ["model"], registry=reg)

    with get_session() as s:
        counts: dict[str, int] = {}
        workers = s.query(Worker).all()
        for w in workers:
            counts[w.state] = counts.get(w.state, 0) + 1
            for d in (w.status or {}).get("gpu_devices", []):
                idx = str(d.get("index", 0))
                g_gpu_util.labels(w.name, idx).set(d.get("core", {}).get("utilization_rate", 0))
                mem = d.get("memory", {})
                g_vram_total.labels(w.name, idx).set(mem.get("total", 0))
                g_vram_used.labels(w.name, idx).set(mem.get("used", 0))
        for st in WorkerState:
            g_workers.labels(st.value).set(counts.get(st.value, 0))
        # allocated = sum of claims
        alloc: dict[tuple[str, str], int] = {}
        wmap = {w.id: w.name for w in workers}
        for inst in s.query(ModelInstance).all():
            wname = wmap.get(inst.worker_id)
            if wname:
                for idx, b in ((inst.computed_resource_claim or {}).get("vram") or {}).items():
                    alloc[(wname, idx)] = alloc.get((wname, idx), 0) + b
        for (wname, idx), b in alloc.items():
            g_vram_alloc.labels(wname, idx).set(b)
        inst_counts: dict[tuple[str, str], int] = {}
        for inst in s.query(ModelInstance).all():
            key = (inst.model_name, inst.state)
            inst_counts[key] = inst_counts.get(key, 0) + 1
        for (mname, st), n in inst_counts.items():
            g_instances.labels(mname, st).set(n)
        for m in s.query(Model).all():
            g_replicas.labels(m.name).set(m.replicas)
        usage_by_model: dict[str, list[int]] = {}
        for u in s.query(ModelUsage).all():
            agg = usage_by_model.setdefault(u.model_name, [0, 0, 0])
            agg[0] += u.prompt_tokens
            agg[1] += u.completion_tokens
            agg[2] += u.request_count
        for mname, (p, c, r) in usage_by_model.items():
            g_prompt.labels(mname).set(p)
            g_completion.labels(mname).set(c)
            g_requests.labels(mname).set(r)
    return generate_latest(reg)
