"""Rate-limited work queue (reference: gpustack/server/workqueue.py:94,130):
coalescing, exponential backoff, dirty re-adds."""
import threading
import time

from gpustack_amd.server.workqueue import ExponentialBackoff, WorkQueue


def test_coalescing_adds():
    wq = WorkQueue()
    wq.add("a")
    wq.add("a")
    wq.add("b")
    assert len(wq) == 2
    assert wq.get(timeout=1) == "a"
    assert wq.get(timeout=1) == "b"
    assert wq.get(timeout=0.05) is None


def test_backoff_growth_and_reset():
    bo = ExponentialBackoff(base=1.0, cap=8.0)
    assert [bo.next_delay("x") for _ in range(5)] == [1, 2, 4, 8, 8]
    bo.forget("x")
    assert bo.next_delay("x") == 1


def test_requeue_applies_delay():
    wq = WorkQueue(base_delay=0.2, max_delay=1.0)
    wq.add("a")
    item = wq.get(timeout=1)
    t0 = time.monotonic()
    wq.done(item, requeue=True)
    assert wq.get(timeout=0.05) is None      # not due yet
    assert wq.get(timeout=2) == "a"          # due after the backoff
    assert time.monotonic() - t0 >= 0.18
    wq.done("a")                             # success resets the counter
    assert wq.backoff.failures.get("a") is None


def test_dirty_readd_while_processing():
    wq = WorkQueue()
    wq.add("a")
    item = wq.get(timeout=1)
    wq.add("a")                              # arrives mid-processing
    assert wq.get(timeout=0.05) is None      # not queued twice concurrently
    wq.done(item)
    assert wq.get(timeout=1) == "a"          # coalesced re-add fires


def test_get_blocks_until_add():
    wq = WorkQueue()
    got = []

    def consumer():
        got.append(wq.get(timeout=5))

    t = threading.Thread(target=consumer)
    t.start()
    time.sleep(0.1)
    wq.add("late")
    t.join(timeout=5)
    assert got == ["late"]


def test_shutdown_unblocks():
    wq = WorkQueue()

    def stopper():
        time.sleep(0.1)
        wq.shutdown()

    threading.Thread(target=stopper).start()
    assert wq.get(timeout=5) is None
