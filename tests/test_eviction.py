"""Node-pressure eviction (kubelet memory.available hard-eviction
analogue — the reference has no local pods and so no eviction at all).
Ranking follows k8s: BestEffort → Burstable → Guaranteed, lower priority
first, bigger memory usage first; critical priority exempt; evicted pods
end Failed/Evicted and are NOT deleted."""

import time

import pytest

from k8s_runpod_kubelet_amd.provider.eviction import (
    CRITICAL_PRIORITY,
    EvictionManager,
    rank_victims,
)
from tests.conftest import make_pod, wait_until


def _pod(name, qos="besteffort", priority=0):
    pod = make_pod(name)
    pod["spec"]["priority"] = priority
    c = pod["spec"]["containers"][0]
    if qos == "guaranteed":
        c["resources"] = {"requests": {"cpu": "1", "memory": "1Gi"},
                          "limits": {"cpu": "1", "memory": "1Gi"}}
    elif qos == "burstable":
        c["resources"] = {"requests": {"memory": "256Mi"}}
    return pod


def test_rank_victims_qos_order():
    cands = [("g", _pod("g", "guaranteed"), 100),
             ("be", _pod("be", "besteffort"), 100),
             ("bu", _pod("bu", "burstable"), 100)]
    assert [k for k, _, _ in rank_victims(cands)] == ["be", "bu", "g"]


def test_rank_victims_priority_then_usage():
    cands = [("low-small", _pod("a", priority=0), 10),
             ("low-big", _pod("b", priority=0), 1000),
             ("high", _pod("c", priority=100), 5000)]
    ranked = [k for k, _, _ in rank_victims(cands)]
    assert ranked == ["low-big", "low-small", "high"]


def test_rank_victims_critical_exempt():
    cands = [("crit", _pod("crit", priority=CRITICAL_PRIORITY), 10**9),
             ("norm", _pod("norm"), 10)]
    assert [k for k, _, _ in rank_victims(cands)] == ["norm"]


class FakeProvider:
    def __init__(self, cands):
        self.cands = cands
        self.evicted = []

    def eviction_candidates(self):
        return self.cands

    def evict_pod(self, key, message):
        self.evicted.append((key, message))


def test_manager_threshold_and_single_eviction():
    prov = FakeProvider([("p1", _pod("p1"), 100),
                         ("p2", _pod("p2", "guaranteed"), 100)])
    avail = [10 << 20]  # 10 MiB available
    mgr = EvictionManager(prov, threshold_bytes=100 << 20,
                          meminfo_reader=lambda: avail[0])
    assert mgr.check() == ["p1"]  # one victim per pass, BestEffort first
    avail[0] = 200 << 20  # pressure relieved
    assert mgr.check() == []
    assert len(prov.evicted) == 1
    assert "low on resource: memory" in prov.evicted[0][1]


def test_manager_disabled_and_unreadable_signal():
    prov = FakeProvider([("p1", _pod("p1"), 100)])
    assert EvictionManager(prov, 0, lambda: 0).check() == []
    assert EvictionManager(prov, 100, lambda: -1).check() == []
    assert prov.evicted == []


def test_eviction_end_to_end_stack(tmp_state_dir):
    """Full stack with a tiny eviction interval and an injected pressure
    signal: the BestEffort pod is killed, its object goes Failed/Evicted
    and STAYS (not deleted); the Guaranteed pod survives."""
    from k8s_runpod_kubelet_amd.app import build_stack
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube

    cfg = Config(state_dir=tmp_state_dir, gpu_count_override=8,
                 runtime="fake", pending_retry_interval_s=0.2,
                 notify_interval_s=0, eviction_memory_threshold_mb=100,
                 eviction_interval_s=0.1)
    kube = FakeKube()
    stack = build_stack(cfg, client=kube)
    pressure = [1 << 40]  # plenty available at start

    stack.provider.meminfo_reader = lambda: pressure[0]
    stack.start(serve_http=False)
    try:
        kube.create_pod("default", _pod("victim", "besteffort"))
        kube.create_pod("default", _pod("survivor", "guaranteed"))
        assert wait_until(
            lambda: all(
                (kube.get_pod("default", n).get("status") or {}).get(
                    "phase") == "Running"
                for n in ("victim", "survivor")), timeout_s=10)

        pressure[0] = 1 << 20  # 1 MiB available: pressure
        assert wait_until(
            lambda: (kube.get_pod("default", "victim").get("status") or
                     {}).get("reason") == "Evicted", timeout_s=10)
        v = kube.get_pod("default", "victim")
        assert v["status"]["phase"] == "Failed"
        assert "low on resource: memory" in v["status"]["message"]
        pressure[0] = 1 << 40  # relieved before the next tick fires again

        # survivor untouched; evicted object still present (not deleted)
        time.sleep(0.5)
        s = kube.get_pod("default", "survivor")
        assert (s.get("status") or {}).get("phase") == "Running"
        assert kube.get_pod("default", "victim") is not None

        # Evicted event surfaced
        with kube._lock:
            evs = [e for e in kube.events.objects.values()
                   if e.get("reason") == "Evicted"]
        assert evs and "memory" in evs[0]["message"]
    finally:
        from tests.conftest import drain_runtime

        drain_runtime(stack.runtime)
        stack.stop()


def test_gpu_failure_evicts_bound_pods(tmp_state_dir):
    """A GPU that loses health mid-run (RAS) fails its bound pods fast
    (reason GPUFailure) so controllers reschedule; pods on healthy GPUs
    survive; the evicted object stays visible."""
    from k8s_runpod_kubelet_amd.app import build_stack
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube

    cfg = Config(state_dir=tmp_state_dir, gpu_count_override=8,
                 runtime="fake", pending_retry_interval_s=0.2,
                 notify_interval_s=0)
    kube = FakeKube()
    stack = build_stack(cfg, client=kube)
    stack.start(serve_http=False)
    try:
        def gpu_pod(name):
            pod = make_pod(name)
            pod["spec"]["containers"][0]["resources"] = {
                "limits": {"amd.com/gpu": "1"}}
            return pod

        kube.create_pod("default", gpu_pod("g1"))
        kube.create_pod("default", gpu_pod("g2"))
        assert wait_until(
            lambda: all(
                (kube.get_pod("default", n).get("status") or {}).get(
                    "phase") == "Running" for n in ("g1", "g2")),
            timeout_s=10)
        bound = {}
        with stack.provider._pods_lock:
            for k, info in stack.provider._pod_status.items():
                bound[k.split("-", 1)[1]] = list(info.gpu_indices)
        assert bound["g1"] and bound["g2"]
        assert bound["g1"] != bound["g2"]

        # GPU of g1 goes unhealthy; reconcile tick notices
        dead = bound["g1"][0]
        for g in stack.inventory.gpus:
            if g.index == dead:
                g.healthy = False
        stack.provider._periodic_reconcile()

        v = kube.get_pod("default", "g1")
        assert (v.get("status") or {}).get("reason") == "GPUFailure"
        assert v["status"]["phase"] == "Failed"
        s = kube.get_pod("default", "g2")
        assert (s.get("status") or {}).get("phase") == "Running"
        with kube._lock:
            assert any(e.get("reason") == "GPUFailure"
                       for e in kube.events.objects.values())
    finally:
        from tests.conftest import drain_runtime

        drain_runtime(stack.runtime)
        stack.stop()


@pytest.mark.skipif(__import__("os").geteuid() != 0,
                    reason="process runtime tests need root")
def test_eviction_kills_real_process_and_frees_gpu(tmp_state_dir):
    """Hard eviction against the real ProcessRuntime: the workload process
    dies, the GPU reservation frees (re-placeable), nothing leaks."""
    import os

    from k8s_runpod_kubelet_amd.app import build_stack
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube

    cfg = Config(state_dir=tmp_state_dir, gpu_count_override=1,
                 pending_retry_interval_s=0.2, notify_interval_s=0,
                 eviction_memory_threshold_mb=100,
                 eviction_interval_s=0.1)
    kube = FakeKube()
    stack = build_stack(cfg, client=kube)
    stack.runtime.enable_cgroups = False
    pressure = [1 << 40]
    stack.provider.meminfo_reader = lambda: pressure[0]
    stack.start(serve_http=False)
    try:
        pod = make_pod("evictme")
        pod["spec"]["containers"][0]["command"] = ["sleep"]
        pod["spec"]["containers"][0]["args"] = ["300"]
        pod["spec"]["containers"][0]["resources"] = {
            "limits": {"amd.com/gpu": "1"}}
        kube.create_pod("default", pod)
        assert wait_until(
            lambda: (kube.get_pod("default", "evictme").get("status") or
                     {}).get("phase") == "Running", timeout_s=10)
        with stack.provider._pods_lock:
            info = next(iter(stack.provider._pod_status.values()))
            iid = info.instance_id
        pid = stack.runtime.get_detailed_status(iid).containers[0].pid
        assert pid and os.path.exists(f"/proc/{pid}")

        pressure[0] = 1 << 20
        assert wait_until(
            lambda: (kube.get_pod("default", "evictme").get("status") or
                     {}).get("reason") == "Evicted", timeout_s=10)
        pressure[0] = 1 << 40
        # process actually dies and the GPU frees for the next pod
        assert wait_until(lambda: not os.path.exists(f"/proc/{pid}")
                          or open(f"/proc/{pid}/stat").read()
                          .rsplit(") ", 1)[-1].startswith("Z"),
                          timeout_s=10)
        assert wait_until(
            lambda: all(not s.pod_keys for s in stack.ledger.snapshot()),
            timeout_s=10)
        # node can place a new GPU pod after the eviction
        pod2 = make_pod("next")
        pod2["spec"]["containers"][0]["command"] = ["sleep"]
        pod2["spec"]["containers"][0]["args"] = ["30"]
        pod2["spec"]["containers"][0]["resources"] = {
            "limits": {"amd.com/gpu": "1"}}
        kube.create_pod("default", pod2)
        assert wait_until(
            lambda: (kube.get_pod("default", "next").get("status") or
                     {}).get("phase") == "Running", timeout_s=10)
    finally:
        from tests.conftest import drain_runtime

        drain_runtime(stack.runtime)
        stack.stop()


def test_memory_pressure_node_condition(tmp_state_dir):
    """node.status MemoryPressure flips True while the eviction signal is
    below threshold (same signal the kubelet uses)."""
    from k8s_runpod_kubelet_amd.app import build_stack
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube

    cfg = Config(state_dir=tmp_state_dir, gpu_count_override=2,
                 runtime="fake", eviction_memory_threshold_mb=100)
    stack = build_stack(cfg, client=FakeKube())
    pressure = [1 << 40]
    stack.provider.meminfo_reader = lambda: pressure[0]

    def cond(node, typ):
        return next(c for c in node["status"]["conditions"]
                    if c["type"] == typ)

    node = stack.provider.get_node_status()
    assert cond(node, "MemoryPressure")["status"] == "False"
    pressure[0] = 1 << 20
    node = stack.provider.get_node_status()
    assert cond(node, "MemoryPressure")["status"] == "True"
    assert cond(node, "MemoryPressure")["reason"] == \
        "KubeletHasInsufficientMemory"
