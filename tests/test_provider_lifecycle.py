"""Hermetic provider lifecycle tests — the suite the reference cannot run
without real cloud credentials (reference runpod_test.go requires
RUNPOD_API_KEY + KUBECONFIG; SURVEY §4 gap)."""

import time

import pytest

from k8s_runpod_kubelet_amd.config import Config
from k8s_runpod_kubelet_amd.kube.client import NotFoundError
from k8s_runpod_kubelet_amd.provider import annotations as ann
from k8s_runpod_kubelet_amd.provider.provider import Provider
from k8s_runpod_kubelet_amd.runtime.fake import FakeRuntime
from k8s_runpod_kubelet_amd.runtime.types import PodStatus
from tests.conftest import make_pod, wait_until


@pytest.fixture
def cfg():
    return Config(pending_retry_interval_s=0.1, reconcile_interval_s=60,
                  cleanup_interval_s=60, notify_interval_s=0)


@pytest.fixture
def setup(fake_kube, cfg):
    rt = FakeRuntime(gpu_count=8)
    prov = Provider(fake_kube, cfg, rt)
    yield fake_kube, rt, prov
    prov.stop()


def create_tracked(kube, prov, pod):
    created = kube.create_pod(pod["metadata"].get("namespace", "default"), pod)
    prov.create_pod(created)
    return created


def test_create_deploys_and_writes_annotations(setup):
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod(gpus=1))
    pod = kube.get_pod("default", "p1")
    assert pod["metadata"]["annotations"][ann.POD_ID].startswith("fake-")
    assert pod["metadata"]["annotations"][ann.GPU_IDS] == "0"
    assert float(pod["metadata"]["annotations"][ann.COST_PER_HR]) > 0
    # status pushed: Running + Ready (no ports requested)
    pod = wait_until(lambda: (
        kube.get_pod("default", "p1")
        if kube.get_pod("default", "p1").get("status", {}).get("phase") == "Running"
        else None))
    assert pod["status"]["phase"] == "Running"
    conds = {c["type"]: c["status"] for c in pod["status"]["conditions"]}
    assert conds["Ready"] == "True"


def test_deploy_failure_leaves_pending_then_retries(setup):
    kube, rt, prov = setup
    rt.deploy_error = "backend exploded"
    create_tracked(kube, prov, make_pod(gpus=1))
    info = prov.instance_info("default", "p1")
    assert info is not None and not info.instance_id  # still pending
    # reference kubelet.go:412-415: CreatePod returns nil, retry loop deploys
    rt.deploy_error = None
    prov.process_pending_pods()
    info = prov.instance_info("default", "p1")
    assert info.instance_id


def test_pending_timeout_marks_failed(fake_kube):
    cfg = Config(pending_pod_timeout_s=0.05, pending_retry_interval_s=999)
    rt = FakeRuntime(deploy_error="always down")
    prov = Provider(fake_kube, cfg, rt)
    create_tracked(fake_kube, prov, make_pod(gpus=1))
    time.sleep(0.1)
    prov.process_pending_pods()
    pod = fake_kube.get_pod("default", "p1")
    assert pod["status"]["phase"] == "Failed"
    assert pod["status"]["reason"] == "DeploymentFailed"
    prov.stop()


def test_completion_success(setup):
    kube, rt, prov = setup
    created = create_tracked(kube, prov, make_pod(gpus=1))
    iid = prov.instance_info("default", "p1").instance_id
    rt.complete_instance(iid)
    prov.update_all_pod_statuses()
    pod = kube.get_pod("default", "p1")
    assert pod["status"]["phase"] == "Succeeded"
    term = pod["status"]["containerStatuses"][0]["state"]["terminated"]
    assert term["exitCode"] == 0


def test_completion_failure(setup):
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod(gpus=1))
    iid = prov.instance_info("default", "p1").instance_id
    rt.fail_instance(iid, exit_code=5, message="boom failure")
    prov.update_all_pod_statuses()
    pod = kube.get_pod("default", "p1")
    assert pod["status"]["phase"] == "Failed"
    assert pod["status"]["containerStatuses"][0]["state"]["terminated"]["exitCode"] == 5


def test_missing_instance_strips_annotations(setup):
    # reference handleMissingRunPodInstance (kubelet.go:1708-1773)
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod(gpus=1))
    iid = prov.instance_info("default", "p1").instance_id
    rt.vanish_instance(iid)
    prov.update_all_pod_statuses()
    pod = kube.get_pod("default", "p1")
    assert pod["status"]["phase"] == "Failed"
    assert pod["status"]["reason"] == "PodDeleted"
    assert ann.POD_ID not in pod["metadata"]["annotations"]
    assert ann.GPU_IDS not in pod["metadata"]["annotations"]


def test_delete_pod_terminates_backend(setup):
    kube, rt, prov = setup
    created = create_tracked(kube, prov, make_pod(gpus=1))
    pod = kube.get_pod("default", "p1")
    prov.delete_pod(pod)
    assert rt.terminate_calls == 1
    assert prov.get_pod("default", "p1") is None
    # deletedPods entry recorded for the cleanup loop
    assert "default/p1" in prov._deleted_pods


def test_cleanup_deleted_pods(setup):
    # reference cleanupDeletedPods (kubelet.go:1190-1227)
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod(gpus=1))
    pod = kube.get_pod("default", "p1")
    prov.delete_pod(pod)
    # pod still exists in K8s -> entry kept
    prov.cleanup_deleted_pods()
    assert "default/p1" in prov._deleted_pods
    kube.delete_pod("default", "p1", grace_period_s=0)
    prov.cleanup_deleted_pods()
    assert "default/p1" not in prov._deleted_pods


def test_stuck_terminating_no_id_force_deleted(setup):
    kube, rt, prov = setup
    kube.create_pod("default", make_pod("stuck"))
    kube.delete_pod("default", "stuck")  # sets deletionTimestamp
    prov.cleanup_stuck_terminating_pods()
    with pytest.raises(NotFoundError):
        kube.get_pod("default", "stuck")


def test_stuck_terminating_backend_dead_force_deleted(setup):
    kube, rt, prov = setup
    pod = make_pod("stuck2", annotations={ann.POD_ID: "fake-gone"})
    kube.create_pod("default", pod)
    kube.delete_pod("default", "stuck2")
    prov.cleanup_stuck_terminating_pods()  # NOT_FOUND on backend => force
    with pytest.raises(NotFoundError):
        kube.get_pod("default", "stuck2")


def test_stuck_terminating_alive_reterminated_then_forced(fake_kube):
    cfg = Config(stuck_reterminate_after_s=0.0, stuck_force_after_s=3600)
    rt = FakeRuntime()
    prov = Provider(fake_kube, cfg, rt)
    st = rt.deploy(__import__(
        "k8s_runpod_kubelet_amd.runtime.types", fromlist=["DeployParams"]
    ).DeployParams(pod_key="default-stuck3", name="stuck3"))
    pod = make_pod("stuck3", annotations={ann.POD_ID: st.id})
    fake_kube.create_pod("default", pod)
    fake_kube.delete_pod("default", "stuck3")
    before = rt.terminate_calls
    prov.cleanup_stuck_terminating_pods()
    assert rt.terminate_calls == before + 1  # re-terminate (kubelet.go:1333)
    fake_kube.get_pod("default", "stuck3")  # still exists (not forced yet)
    prov.stop()


def test_load_running_adopts_and_imports(setup):
    kube, rt, prov = setup
    # a) K8s pod with matching backend instance -> adopted
    from k8s_runpod_kubelet_amd.runtime.types import DeployParams

    st = rt.deploy(DeployParams(pod_key="default-adopt", name="adopt", gpu_count=1))
    kube.create_pod("default", make_pod("adopt", annotations={
        ann.POD_ID: st.id, ann.GPU_IDS: "0"}))
    # b) K8s pod with id but instance missing -> handleMissingRunPodInstance
    kube.create_pod("default", make_pod("orphan-id", annotations={
        ann.POD_ID: "fake-missing"}))
    # c) K8s pod without id -> pending (retry loop)
    kube.create_pod("default", make_pod("no-id"))
    # d) backend instance with no K8s pod -> virtual pod import
    st2 = rt.deploy(DeployParams(pod_key="default-lost", name="lost"))

    prov.load_running()

    assert prov.instance_info("default", "adopt").instance_id == st.id
    orphan = kube.get_pod("default", "orphan-id")
    assert orphan["status"]["phase"] == "Failed"
    info = prov.instance_info("default", "no-id")
    assert info is not None and not info.instance_id
    virt = kube.get_pod("default", f"runpod-{st2.id}")
    assert virt["metadata"]["annotations"][ann.EXTERNAL] == "true"
    assert virt["spec"]["nodeName"] == prov.node_name
    assert virt["status"]["phase"] == "Running"


def test_node_status_capacity_and_taint(fake_kube, cfg, synthetic_ledger):
    rt = FakeRuntime(gpu_count=8)
    prov = Provider(fake_kube, cfg, rt, ledger=synthetic_ledger)
    node = prov.get_node_status()
    assert node["metadata"]["name"] == "virtual-runpod"
    assert node["status"]["capacity"][ann.GPU_RESOURCE] == "8"
    assert node["status"]["allocatable"][ann.GPU_RESOURCE] == "8"
    taints = node["spec"]["taints"]
    assert taints == [{"key": ann.TAINT_KEY, "value": ann.TAINT_VALUE,
                       "effect": "NoSchedule"}]
    assert node["metadata"]["labels"]["type"] == "virtual-kubelet"
    # per-GPU HBM annotations exported for workload schedulers
    assert node["metadata"]["annotations"]["amd.com/gpu-0-hbm-free-bytes"] == str(288 * 1024**3)
    prov.stop()


def test_ping_fails_when_backend_unhealthy(setup):
    kube, rt, prov = setup
    prov.ping()
    rt.set_healthy(False)
    with pytest.raises(RuntimeError):
        prov.ping()
    node = prov.get_node_status()
    ready = [c for c in node["status"]["conditions"] if c["type"] == "Ready"][0]
    assert ready["status"] == "False"


def test_get_pod_status_translates(setup):
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod(gpus=1))
    status = prov.get_pod_status("default", "p1")
    assert status["phase"] == "Running"
    assert prov.get_pod_status("default", "absent") is None


def test_update_pod_refreshes_cache(setup):
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod())
    pod = kube.get_pod("default", "p1")
    pod["metadata"]["labels"]["new"] = "label"
    prov.update_pod(pod)
    assert prov.get_pod("default", "p1")["metadata"]["labels"]["new"] == "label"


def test_datacenter_annotation_injection(fake_kube):
    # reference kubelet.go:437-455: node-level datacenter annotation injected
    cfg = Config(datacenter_ids=["dc-a"])
    rt = FakeRuntime()
    prov = Provider(fake_kube, cfg, rt)
    create_tracked(fake_kube, prov, make_pod(gpus=1))
    pod = fake_kube.get_pod("default", "p1")
    assert pod["metadata"]["annotations"][ann.DATACENTER_IDS] == "dc-a"
    prov.stop()


def test_get_container_logs(setup):
    kube, rt, prov = setup
    create_tracked(kube, prov, make_pod(gpus=1))
    assert "fake logs" in prov.get_container_logs("default", "p1")
    assert prov.get_container_logs("default", "absent") == ""
