import threading

import pytest

from k8s_runpod_kubelet_amd.kube.client import ConflictError, NotFoundError
from tests.conftest import make_pod


def test_pod_crud(fake_kube):
    pod = make_pod("a")
    created = fake_kube.create_pod("default", pod)
    assert created["metadata"]["uid"]
    assert created["metadata"]["resourceVersion"]
    got = fake_kube.get_pod("default", "a")
    assert got["metadata"]["name"] == "a"
    with pytest.raises(ConflictError):
        fake_kube.create_pod("default", make_pod("a"))
    with pytest.raises(NotFoundError):
        fake_kube.get_pod("default", "zzz")


def test_field_selector(fake_kube):
    fake_kube.create_pod("default", make_pod("a", node="n1"))
    fake_kube.create_pod("default", make_pod("b", node="n2"))
    pods = fake_kube.list_pods(field_selector="spec.nodeName=n1")
    assert [p["metadata"]["name"] for p in pods] == ["a"]
    pods = fake_kube.list_pods(field_selector="spec.nodeName!=n1")
    assert [p["metadata"]["name"] for p in pods] == ["b"]


def test_update_conflict_on_stale_rv(fake_kube):
    fake_kube.create_pod("default", make_pod("a"))
    p1 = fake_kube.get_pod("default", "a")
    p2 = fake_kube.get_pod("default", "a")
    p1["metadata"]["labels"]["x"] = "1"
    fake_kube.update_pod("default", p1)
    p2["metadata"]["labels"]["y"] = "2"
    with pytest.raises(ConflictError):
        fake_kube.update_pod("default", p2)


def test_status_patch_merges_conditions(fake_kube):
    fake_kube.create_pod("default", make_pod("a"))
    fake_kube.patch_pod_status("default", "a", {"status": {
        "phase": "Pending",
        "conditions": [{"type": "Ready", "status": "False"}],
    }})
    fake_kube.patch_pod_status("default", "a", {"status": {
        "phase": "Running",
        "conditions": [{"type": "Ready", "status": "True"},
                       {"type": "PodScheduled", "status": "True"}],
    }})
    pod = fake_kube.get_pod("default", "a")
    conds = {c["type"]: c["status"] for c in pod["status"]["conditions"]}
    assert conds == {"Ready": "True", "PodScheduled": "True"}
    assert pod["status"]["phase"] == "Running"


def test_two_phase_delete(fake_kube):
    fake_kube.create_pod("default", make_pod("a"))
    fake_kube.delete_pod("default", "a")  # graceful: sets deletionTimestamp
    pod = fake_kube.get_pod("default", "a")
    assert pod["metadata"]["deletionTimestamp"]
    fake_kube.delete_pod("default", "a", grace_period_s=0)  # finalize
    with pytest.raises(NotFoundError):
        fake_kube.get_pod("default", "a")


def test_watch_delivers_events(fake_kube):
    events = []
    done = threading.Event()

    def watcher():
        for ev_type, pod in fake_kube.watch_pods(field_selector="spec.nodeName=n1",
                                                 timeout_s=5):
            events.append((ev_type, pod["metadata"]["name"]))
            if ev_type == "DELETED":
                done.set()
                return

    t = threading.Thread(target=watcher, daemon=True)
    t.start()
    import time
    time.sleep(0.1)
    fake_kube.create_pod("default", make_pod("w1", node="n1"))
    fake_kube.create_pod("default", make_pod("other", node="n2"))  # filtered out
    fake_kube.patch_pod("default", "w1", {"metadata": {"labels": {"x": "1"}}})
    fake_kube.delete_pod("default", "w1", grace_period_s=0)
    assert done.wait(5)
    names = [n for _, n in events]
    assert "other" not in names
    assert [t for t, _ in events] == ["ADDED", "MODIFIED", "DELETED"]


def test_annotation_patch_with_none_deletes(fake_kube):
    fake_kube.create_pod("default", make_pod("a", annotations={"k1": "v1", "k2": "v2"}))
    fake_kube.patch_pod("default", "a", {"metadata": {"annotations": {"k1": None}}})
    pod = fake_kube.get_pod("default", "a")
    assert "k1" not in pod["metadata"]["annotations"]
    assert pod["metadata"]["annotations"]["k2"] == "v2"


def test_node_and_lease(fake_kube):
    fake_kube.create_node({"metadata": {"name": "n1"}, "status": {}})
    fake_kube.patch_node_status("n1", {"status": {"capacity": {"cpu": "4"}}})
    assert fake_kube.get_node("n1")["status"]["capacity"]["cpu"] == "4"
    lease = {"metadata": {"name": "n1"}, "spec": {"holderIdentity": "n1"}}
    fake_kube.create_lease("kube-node-lease", lease)
    got = fake_kube.get_lease("kube-node-lease", "n1")
    got["spec"]["renewTime"] = "now"
    fake_kube.update_lease("kube-node-lease", got)
    fake_kube.set_leases_supported(False)
    assert not fake_kube.leases_supported()
