"""Benchmark SLA + saturation analysis (reference: SLA_THRESHOLDS,
schemas/benchmark.py:104-197; saturation probe in the benchmark runner)."""
from gpustack_amd.bench.analysis import analyze_profile, sla_check


def _pt(value, tps, ttft_p99=100.0, tpot_p99=20.0, failed=0, total=100):
    return {"value": value, "output_tps": tps, "ttft_p99_ms": ttft_p99,
            "tpot_p99_ms": tpot_p99, "requests": total,
            "failed_requests": failed}


def test_sla_check_passes_and_fails():
    assert sla_check(_pt(1, 100))["passed"]
    v = sla_check(_pt(1, 100, ttft_p99=9000.0))
    assert not v["passed"]
    assert v["violations"][0]["metric"] == "ttft_p99_ms"
    # error-rate violation derived from request counts
    v = sla_check(_pt(1, 100, failed=5))
    assert not v["passed"]
    assert v["violations"][0]["metric"] == "error_rate"
    # custom thresholds override the defaults
    assert not sla_check(_pt(1, 100, ttft_p99=150.0),
                         {"ttft_p99_ms": 100.0})["passed"]


def test_analyze_profile_saturation_and_sla_knee():
    profile = [
        _pt(8, 1000.0),
        _pt(16, 1900.0),
        _pt(32, 3500.0),
        _pt(64, 3600.0, ttft_p99=6000.0),   # saturated AND out of SLA
        _pt(128, 3620.0, ttft_p99=20000.0),
    ]
    a = analyze_profile(profile)
    assert a["max_sla_load"] == 32          # last point inside SLA
    assert a["saturation_load"] == 64       # <5% tps gain from 32 -> 64
    assert a["peak_output_tps"] == 3620.0 and a["peak_load"] == 128
    verdicts = {v["value"]: v["passed"] for v in a["sla"]}
    assert verdicts == {8: True, 16: True, 32: True, 64: False, 128: False}


def test_analyze_single_point():
    a = analyze_profile([_pt(8, 500.0)])
    assert a["max_sla_load"] == 8
    assert a["saturation_load"] is None
