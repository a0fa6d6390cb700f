"""HA leader election over DB leases."""
import tempfile
import time

from gpustack_amd.config import Config
from gpustack_amd.db import init_db
from gpustack_amd.server.coordinator import LeaseCoordinator


def test_single_leader_and_takeover():
    cfg = Config(data_dir=tempfile.mkdtemp())
    cfg.ensure_dirs()
    init_db(cfg.resolved_database_url())

    a = LeaseCoordinator(ttl=1.0, holder="node-a")
    b = LeaseCoordinator(ttl=1.0, holder="node-b")
    assert a.try_acquire()
    assert not b.try_acquire()
    assert a.is_leader and not b.is_leader
    # renewal keeps leadership
    time.sleep(0.5)
    assert a.try_acquire()
    assert not b.try_acquire()
    # leader stops renewing -> lease expires -> takeover
    time.sleep(1.2)
    assert b.try_acquire()
    assert b.is_leader
    # a notices loss on next attempt
    lost = []
    a.on_lost = lambda: lost.append(1)
    assert not a.try_acquire()
    assert not a.is_leader and lost
    # release clears the row
    b.release()
    c = LeaseCoordinator(ttl=1.0, holder="node-c")
    assert c.try_acquire()


def test_leader_gating_blocks_follower_controllers():
    from gpustack_amd.server.controllers import ModelController

    cfg = Config(data_dir=tempfile.mkdtemp())
    cfg.ensure_dirs()
    init_db(cfg.resolved_database_url())

    class FakeCoord:
        is_leader = False

    mc = ModelController(cfg)
    mc.coordinator = FakeCoord()
    assert not mc._is_leader()
    FakeCoord.is_leader = True
    assert mc._is_leader()
