"""GPU + system detection for MI355X workers.

Replaces gpustack-runtime vendor detection + fastfetch (SURVEY.md §2.9
#3/#4) with direct amdsmi / rocm-smi / torch probing and /proc readers —
no external binaries. Static `gpu_devices` config overrides detection for
air-gapped setups (reference: config.get_gpu_devices, config.py:623).
"""
from __future__ import annotations

import logging
import os
import platform
import shutil
import subprocess

logger = logging.getLogger(__name__)

MI355X_VRAM = 288 * 1024**3


def _detect_amdsmi() -> list[dict] | None:
    try:
        import amdsmi  # type: ignore

        amdsmi.amdsmi_init()
        try:
            devs = []
            for i, h in enumerate(amdsmi.amdsmi_get_processor_handles()):
                info = amdsmi.amdsmi_get_gpu_asic_info(h)
                mem = amdsmi.amdsmi_get_gpu_memory_total(h, amdsmi.AmdSmiMemoryType.VRAM)
                used = amdsmi.amdsmi_get_gpu_memory_usage(h, amdsmi.AmdSmiMemoryType.VRAM)
                try:
                    act = amdsmi.amdsmi_get_gpu_activity(h).get("gfx_activity", 0)
                except Exception:  # noqa: BLE001
                    act = 0
                try:
                    temp = amdsmi.amdsmi_get_temp_metric(
                        h, amdsmi.AmdSmiTemperatureType.JUNCTION,
                        amdsmi.AmdSmiTemperatureMetric.CURRENT)
                except Exception:  # noqa: BLE001
                    temp = 0
                entry = _gpu_entry(
                    index=i,
                    name=info.get("market_name", "AMD Instinct"),
                    uuid=str(info.get("asic_serial", f"amd-{i}")),
                    total=mem, used=used, util=act, temperature=temp,
                )
                # AMD compute/memory partitioning (SPX/DPX/CPX, NPS1/NPS4)
                # — the MI355X-native device-class the scheduler's
                # gpu_type_selector matches on (reference: vGPU slice /
                # MIG-partition classes, schemas/models.py:92-175)
                part = {}
                try:
                    part["compute"] = str(
                        amdsmi.amdsmi_get_gpu_compute_partition(h))
                except Exception:  # noqa: BLE001
                    pass
                try:
                    part["memory"] = str(
                        amdsmi.amdsmi_get_gpu_memory_partition(h))
                except Exception:  # noqa: BLE001
                    pass
                if part:
                    entry["partition"] = part
                devs.append(entry)
            return devs
        finally:
            amdsmi.amdsmi_shut_down()
    except Exception:  # noqa: BLE001
        return None


def _detect_rocm_smi() -> list[dict] | None:
    exe = shutil.which("rocm-smi")
    if not exe:
        return None
    try:
        import json as _json

        out = subprocess.run(
            [exe, "--showmeminfo", "vram", "--showuse", "--showtemp", "--json"],
            capture_output=True, text=True, timeout=20,
        )
        data = _json.loads(out.stdout)
        devs = []
        for key, val in sorted(data.items()):
            if not key.startswith("card"):
                continue
            idx = int(key[4:])
            total = int(val.get("VRAM Total Memory (B)", MI355X_VRAM))
            used = int(val.get("VRAM Total Used Memory (B)", 0))
            util = float(val.get("GPU use (%)", 0) or 0)
            temp = float(val.get("Temperature (Sensor junction) (C)", 0) or 0)
            devs.append(_gpu_entry(index=idx, name="AMD Instinct MI355X",
                                   uuid=f"rocm-{idx}", total=total, used=used,
                                   util=util, temperature=temp))
        return devs or None
    except Exception:  # noqa: BLE001
        return None


def _detect_torch() -> list[dict] | None:
    try:
        import torch

        if not torch.cuda.is_available():
            return None
        devs = []
        for i in range(torch.cuda.device_count()):
            props = torch.cuda.get_device_properties(i)
            free, total = torch.cuda.mem_get_info(i)
            devs.append(_gpu_entry(index=i, name=props.name, uuid=f"torch-{i}",
                                   total=total, used=total - free, util=0.0,
                                   temperature=0.0))
        return devs
    except Exception:  # noqa: BLE001
        return None


def _gpu_entry(index: int, name: str, uuid: str, total: int, used: int,
               util: float, temperature: float) -> dict:
    return {
        "uuid": uuid,
        "name": name,
        "vendor": "AMD",
        "index": index,
        "device_index": index,
        "device_chip_index": 0,
        "arch_family": "gfx950",
        "compute_capability": "gfx950",
        "type": "rocm",
        "core": {"total": 256, "utilization_rate": util},
        "memory": {"total": total, "used": used, "allocated": 0,
                   "is_unified_memory": False},
        "temperature": temperature,
    }


def detect_gpus(static_override: list[dict] | None = None) -> list[dict]:
    if static_override:
        return [_gpu_entry(
            index=d.get("index", i),
            name=d.get("name", "AMD Instinct MI355X"),
            uuid=d.get("uuid", f"static-{i}"),
            total=d.get("memory", {}).get("total", MI355X_VRAM),
            used=d.get("memory", {}).get("used", 0),
            util=0.0, temperature=0.0,
        ) for i, d in enumerate(static_override)]
    for fn in (_detect_amdsmi, _detect_rocm_smi, _detect_torch):
        devs = fn()
        if devs:
            return devs
    return []


def collect_system_status(static_gpus: list[dict] | None = None) -> dict:
    """System + GPU status payload (replaces fastfetch: direct /proc)."""
    status: dict = {
        "os": {"name": platform.system(), "version": platform.release()},
        "kernel": {"release": platform.release(), "architecture": platform.machine()},
    }
    try:
        with open("/proc/meminfo") as f:
            mem = {}
            for line in f:
                parts = line.split(":")
                if parts[0] in ("MemTotal", "MemAvailable", "SwapTotal", "SwapFree"):
                    mem[parts[0]] = int(parts[1].strip().split()[0]) * 1024
        status["memory"] = {
            "total": mem.get("MemTotal", 0),
            "used": mem.get("MemTotal", 0) - mem.get("MemAvailable", 0),
        }
        status["swap"] = {
            "total": mem.get("SwapTotal", 0),
            "used": mem.get("SwapTotal", 0) - mem.get("SwapFree", 0),
        }
    except OSError:
        pass
    try:
        status["cpu"] = {
            "total": os.cpu_count() or 0,
            "utilization_rate": os.getloadavg()[0] / max(1, os.cpu_count() or 1) * 100,
        }
    except OSError:
        pass
    try:
        st = os.statvfs("/")
        status["filesystem"] = [{
            "mount_point": "/",
            "total": st.f_blocks * st.f_frsize,
            "used": (st.f_blocks - st.f_bfree) * st.f_frsize,
        }]
    except OSError:
        pass
    status["gpu_devices"] = detect_gpus(static_gpus)
    return status
