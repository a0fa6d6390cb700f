"""Engine resource fencing via cgroups (reference: gpustack-runtime runs
every engine in a container with resource limits, worker/backends/base.py
mounts/env/limits — SURVEY.md §2.6). MI355X deployments here launch
engines as subprocesses; this module gives those subprocesses the
container-grade resource fences (memory ceiling, CPU quota, pid cap)
without a container runtime: a cgroup per model instance, v2 (unified)
or v1 (split controllers), chosen by what the host mounts.

Best-effort by design: a worker without cgroup write access (non-root,
locked-down delegation) logs one warning and serves unfenced — resource
fencing must never take down serving. GPU isolation itself is
HIP_VISIBLE_DEVICES (serve_manager), which ROCm enforces per-process.
"""
from __future__ import annotations

import logging
import os
from pathlib import Path

logger = logging.getLogger(__name__)

CGROUP_ROOT = "/sys/fs/cgroup"


def _is_v2(root: str = CGROUP_ROOT) -> bool:
    return (Path(root) / "cgroup.controllers").exists()


class CgroupFence:
    """One cgroup (or one per controller on v1) for one engine instance."""

    def __init__(self, name: str, memory_gb: float | None = None,
                 cpus: float | None = None, max_pids: int | None = None,
                 root: str = CGROUP_ROOT):
        self.name = f"gpustack-amd-{name}"
        self.memory_gb = memory_gb
        self.cpus = cpus
        self.max_pids = max_pids
        self.root = root
        self.v2 = _is_v2(root)
        self._dirs: list[Path] = []
        self.active = False

    # -- setup ---------------------------------------------------------------
    def create(self) -> bool:
        try:
            if self.v2:
                self._create_v2()
            else:
                self._create_v1()
            self.active = True
            return True
        except OSError as e:
            logger.warning("resource fence unavailable (%s); engine %s runs "
                           "unfenced", e, self.name)
            self._dirs.clear()
            return False

    def _write(self, path: Path, value: str) -> None:
        path.write_text(value)

    def _create_v2(self) -> None:
        d = Path(self.root) / self.name
        d.mkdir(exist_ok=True)
        self._dirs.append(d)
        if self.memory_gb:
            self._write(d / "memory.max",
                        str(int(self.memory_gb * (1 << 30))))
        if self.cpus:
            period = 100000
            self._write(d / "cpu.max", f"{int(self.cpus * period)} {period}")
        if self.max_pids:
            self._write(d / "pids.max", str(self.max_pids))

    def _create_v1(self) -> None:
        if self.memory_gb:
            d = Path(self.root) / "memory" / self.name
            d.mkdir(exist_ok=True)
            self._dirs.append(d)
            self._write(d / "memory.limit_in_bytes",
                        str(int(self.memory_gb * (1 << 30))))
        if self.cpus:
            d = Path(self.root) / "cpu" / self.name
            d.mkdir(exist_ok=True)
            self._dirs.append(d)
            period = 100000
            self._write(d / "cpu.cfs_period_us", str(period))
            self._write(d / "cpu.cfs_quota_us", str(int(self.cpus * period)))
        if self.max_pids:
            d = Path(self.root) / "pids" / self.name
            d.mkdir(exist_ok=True)
            self._dirs.append(d)
            self._write(d / "pids.max", str(self.max_pids))

    # -- membership ----------------------------------------------------------
    def attach(self, pid: int) -> None:
        """Move the engine process (and its future children) into the
        fence. Must run right after spawn, before the model loads."""
        if not self.active:
            return
        fname = "cgroup.procs" if self.v2 else "tasks"
        for d in self._dirs:
            try:
                self._write(d / fname, str(pid))
            except OSError as e:
                logger.warning("could not attach pid %d to %s: %s", pid, d, e)

    def procs(self) -> list[int]:
        out: set[int] = set()
        fname = "cgroup.procs" if self.v2 else "tasks"
        for d in self._dirs:
            try:
                out.update(int(x) for x in
                           (d / fname).read_text().split())
            except (OSError, ValueError):
                pass
        return sorted(out)

    # -- teardown ------------------------------------------------------------
    def cleanup(self) -> None:
        """Remove the cgroup dirs (after the instance's processes exit —
        rmdir fails while populated; best-effort)."""
        for d in self._dirs:
            try:
                d.rmdir()
            except OSError:
                pass
        self._dirs.clear()
        self.active = False


def fence_from_backend_parameters(instance_name: str, bp: dict) -> CgroupFence | None:
    """backend_parameters knobs (reference analog: container resource
    requests on the runner deployment): memory_limit_gb, cpu_limit,
    pids_limit."""
    mem = bp.get("memory_limit_gb")
    cpus = bp.get("cpu_limit")
    pids = bp.get("pids_limit")
    if not (mem or cpus or pids):
        return None
    try:
        return CgroupFence(instance_name,
                           memory_gb=float(mem) if mem else None,
                           cpus=float(cpus) if cpus else None,
                           max_pids=int(pids) if pids else None)
    except (TypeError, ValueError) as e:
        logger.warning("bad resource-limit backend_parameters (%s); "
                       "ignoring", e)
        return None
