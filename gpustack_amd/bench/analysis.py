"""Benchmark analysis: SLA evaluation + saturation detection over load
sweeps (reference: SLA_THRESHOLDS in gpustack/schemas/benchmark.py:104-197
and the runner's saturation probe, worker/benchmark/analysis.py)."""
from __future__ import annotations

# Default serving SLOs (same metric names the load generator emits).
DEFAULT_SLA = {
    "ttft_p99_ms": 5000.0,
    "tpot_p99_ms": 200.0,
    "error_rate": 0.01,
}

SATURATION_GAIN = 0.05  # <5% extra tok/s from a higher load point = saturated


def sla_check(point: dict, thresholds: dict | None = None) -> dict:
    """{passed, violations: [{metric, value, threshold}]} for one point."""
    thresholds = {**DEFAULT_SLA, **(thresholds or {})}
    violations = []
    total = point.get("requests") or 0
    failed = point.get("failed_requests") or 0
    derived = dict(point, error_rate=(failed / total if total else 0.0))
    for metric, limit in thresholds.items():
        value = derived.get(metric)
        if value is None:
            continue
        if value > limit:
            violations.append({"metric": metric, "value": value,
                               "threshold": limit})
    return {"passed": not violations, "violations": violations}


def analyze_profile(profile: list[dict],
                    thresholds: dict | None = None) -> dict:
    """Sweep-level analysis: per-point SLA verdicts, the highest load still
    inside SLA, the saturation point (where extra load stops buying
    throughput), and the peak throughput point."""
    points = sorted(profile, key=lambda p: p.get("value", 0))
    verdicts = []
    max_sla_load = None
    for p in points:
        v = sla_check(p, thresholds)
        verdicts.append({"value": p.get("value"), **v})
        if v["passed"]:
            max_sla_load = p.get("value")
    saturation_load = None
    for prev, cur in zip(points, points[1:]):
        a, b = prev.get("output_tps") or 0.0, cur.get("output_tps") or 0.0
        if a > 0 and (b - a) / a < SATURATION_GAIN:
            saturation_load = cur.get("value")
            break
    peak = max(points, key=lambda p: p.get("output_tps") or 0.0,
               default=None)
    return {
        "sla": verdicts,
        "max_sla_load": max_sla_load,
        "saturation_load": saturation_load,
        "peak_output_tps": peak.get("output_tps") if peak else None,
        "peak_load": peak.get("value") if peak else None,
    }
