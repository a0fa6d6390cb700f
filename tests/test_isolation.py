"""Engine resource fencing (worker/isolation.py): cgroup v1/v2 layout
written correctly against a fake root, limits parsing from
backend_parameters, and (where the host allows) a live-kernel smoke."""
import os
import subprocess
from pathlib import Path

import pytest

from gpustack_amd.worker.isolation import (
    CgroupFence, fence_from_backend_parameters,
)


def test_v2_layout(tmp_path):
    (tmp_path / "cgroup.controllers").write_text("cpu memory pids")
    f = CgroupFence("i1", memory_gb=2.0, cpus=1.5, max_pids=256,
                    root=str(tmp_path))
    assert f.v2
    assert f.create()
    d = tmp_path / "gpustack-amd-i1"
    assert (d / "memory.max").read_text() == str(2 * (1 << 30))
    assert (d / "cpu.max").read_text() == "150000 100000"
    assert (d / "pids.max").read_text() == "256"
    (d / "cgroup.procs").write_text("")  # kernel-provided in real cgroupfs
    f.attach(1234)
    assert (d / "cgroup.procs").read_text() == "1234"
    # on real cgroupfs the limit files are kernel-virtual and rmdir works;
    # on this fake root clear them first to let cleanup() rmdir
    for child in d.iterdir():
        child.unlink()
    f.cleanup()
    assert not d.exists()


def test_v1_layout(tmp_path):
    for ctl in ("memory", "cpu", "pids"):
        (tmp_path / ctl).mkdir()
    f = CgroupFence("i2", memory_gb=1.0, cpus=2.0, max_pids=64,
                    root=str(tmp_path))
    assert not f.v2
    assert f.create()
    assert (tmp_path / "memory/gpustack-amd-i2/memory.limit_in_bytes"
            ).read_text() == str(1 << 30)
    assert (tmp_path / "cpu/gpustack-amd-i2/cpu.cfs_quota_us"
            ).read_text() == "200000"
    assert (tmp_path / "pids/gpustack-amd-i2/pids.max").read_text() == "64"
    f.attach(77)
    assert (tmp_path / "memory/gpustack-amd-i2/tasks").read_text() == "77"
    for ctl in ("memory", "cpu", "pids"):
        for child in (tmp_path / ctl / "gpustack-amd-i2").iterdir():
            child.unlink()
    f.cleanup()
    assert not (tmp_path / "memory/gpustack-amd-i2").exists()


def test_unwritable_root_is_best_effort(tmp_path):
    f = CgroupFence("i3", memory_gb=1.0, root=str(tmp_path / "nope"))
    assert not f.create()  # warns, returns False
    assert not f.active
    f.attach(1)  # no-op, no raise
    f.cleanup()


def test_fence_from_backend_parameters():
    assert fence_from_backend_parameters("x", {}) is None
    f = fence_from_backend_parameters("x", {"memory_limit_gb": 4,
                                            "cpu_limit": "2",
                                            "pids_limit": 100})
    assert f.memory_gb == 4.0 and f.cpus == 2.0 and f.max_pids == 100
    assert fence_from_backend_parameters("x", {"cpu_limit": "lots"}) is None


@pytest.mark.skipif(
    not os.access("/sys/fs/cgroup", os.W_OK),
    reason="cgroup root not writable (needs root/delegation)")
def test_live_cgroup_attach():
    """Against the real kernel: spawn a process, fence it, verify
    membership through the kernel's procs file, tear down."""
    f = CgroupFence("pytest-live", max_pids=32)
    if not f.create():
        pytest.skip("cgroup hierarchy refused creation")
    p = subprocess.Popen(["sleep", "5"])
    try:
        f.attach(p.pid)
        assert p.pid in f.procs()
    finally:
        p.terminate()
        p.wait()
        f.cleanup()
    # dirs removed (kernel allows rmdir once empty)
    for d in ([Path("/sys/fs/cgroup/gpustack-amd-pytest-live")] +
              [Path(f"/sys/fs/cgroup/{c}/gpustack-amd-pytest-live")
               for c in ("memory", "cpu", "pids")]):
        assert not d.exists()
