"""Minimal 5-field cron matcher (for model scaling schedules —
reference: gpustack/schemas/models.py:237-321 scaling_schedule windows)."""
from __future__ import annotations

import time


def _parse_field(field: str, lo: int, hi: int) -> set[int]:
    out: set[int] = set()
    for part in field.split(","):
        step = 1
        if "/" in part:
            part, step_s = part.split("/")
            step = int(step_s)
        if part in ("*", ""):
            rng = range(lo, hi + 1)
        elif "-" in part:
            a, b = part.split("-")
            rng = range(int(a), int(b) + 1)
        else:
            rng = range(int(part), int(part) + 1)
        out.update(x for x in rng if (x - lo) % step == 0 or step == 1)
    return out


def cron_matches(expr: str, t: float | None = None) -> bool:
    """True if the 5-field cron expression matches the given minute."""
    tm = time.localtime(t if t is not None else time.time())
    fields = expr.split()
    if len(fields) != 5:
        raise ValueError(f"bad cron expression: {expr!r}")
    minute, hour, dom, month, dow = fields
    # cron dow: 0=Sunday..6=Saturday (7 also Sunday); tm_wday: 0=Monday
    cron_dow = (tm.tm_wday + 1) % 7
    checks = [
        (minute, tm.tm_min, 0, 59, False),
        (hour, tm.tm_hour, 0, 23, False),
        (dom, tm.tm_mday, 1, 31, False),
        (month, tm.tm_mon, 1, 12, False),
        (dow, cron_dow, 0, 7, True),
    ]
    for expr_f, val, lo, hi, is_dow in checks:
        if expr_f == "*":
            continue
        allowed = _parse_field(expr_f, lo, hi)
        if is_dow and 7 in allowed:
            allowed.add(0)
        if val not in allowed:
            return False
    return True


def window_active(cron_start: str, duration_minutes: int, t: float | None = None) -> bool:
    """True if `t` is within [fire, fire + duration) for any recent fire
    of the cron expression (checked minute by minute)."""
    now = t if t is not None else time.time()
    for back in range(duration_minutes):
        if cron_matches(cron_start, now - back * 60):
            return True
    return False
