import threading
import time

from k8s_runpod_kubelet_amd.kube.workqueue import WorkQueue


def test_dedup():
    q = WorkQueue()
    q.add("a")
    q.add("a")
    q.add("b")
    assert q.get(0.1) == "a"
    assert q.get(0.1) == "b"
    assert q.get(0.05) is None


def test_dirty_readd_while_processing():
    q = WorkQueue()
    q.add("a")
    key = q.get(0.1)
    q.add("a")  # re-add while processing -> marked dirty
    assert q.get(0.05) is None  # not delivered yet
    q.done(key)
    assert q.get(0.5) == "a"  # redelivered after done


def test_rate_limited_backoff():
    q = WorkQueue(base_delay_s=0.05)
    q.add_rate_limited("a")
    t0 = time.monotonic()
    assert q.get(2.0) == "a"
    assert time.monotonic() - t0 >= 0.04
    q.done("a")
    q.forget("a")


def test_shutdown_unblocks_getters():
    q = WorkQueue()
    results = []

    def getter():
        results.append(q.get(5.0))

    t = threading.Thread(target=getter)
    t.start()
    time.sleep(0.05)
    q.shutdown()
    t.join(2.0)
    assert results == [None]
