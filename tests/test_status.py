"""Status-translation fidelity tests (reference translateRunPodStatus,
kubelet.go:1848-2024 semantics)."""

import time

from k8s_runpod_kubelet_amd.provider.instance import InstanceInfo
from k8s_runpod_kubelet_amd.provider.status import (
    merge_container_status,
    translate_status,
    translate_status_to_phase,
)
from k8s_runpod_kubelet_amd.runtime.types import (
    ContainerRuntimeInfo,
    DetailedStatus,
    PodStatus,
)
from tests.conftest import make_pod


def detailed(status, exit_code=None, message="", ports=None):
    c = ContainerRuntimeInfo(name="main", pid=123, started_at=time.time(),
                             exit_code=exit_code, message=message,
                             ready=status == PodStatus.RUNNING)
    if exit_code is not None:
        c.finished_at = time.time()
    return DetailedStatus(id="i-1", desired_status=status,
                          port_mappings=ports or {}, containers=[c])


def conds(status):
    return {c["type"]: c["status"] for c in status["conditions"]}


def test_phase_map():
    # reference translateRunPodStatusToPhase (kubelet.go:978-995)
    assert translate_status_to_phase(PodStatus.RUNNING) == "Running"
    assert translate_status_to_phase(PodStatus.STARTING) == "Pending"
    assert translate_status_to_phase(PodStatus.EXITED) == "Succeeded"
    assert translate_status_to_phase(PodStatus.TERMINATING) == "Running"
    assert translate_status_to_phase(PodStatus.TERMINATED) == "Succeeded"
    assert translate_status_to_phase(PodStatus.NOT_FOUND) == "Unknown"
    assert translate_status_to_phase("???") == "Unknown"


def test_running_ready_when_ports_exposed():
    pod = make_pod()
    info = InstanceInfo(status=PodStatus.RUNNING, ports_exposed=True)
    st = translate_status(pod, info, detailed(PodStatus.RUNNING), "10.1.2.3")
    assert st["phase"] == "Running"
    assert conds(st)["Ready"] == "True"
    assert conds(st)["ContainersReady"] == "True"
    assert st["hostIP"] == "10.1.2.3"
    cs = st["containerStatuses"][0]
    assert cs["ready"] is True
    assert "running" in cs["state"]
    assert cs["containerID"].startswith("amdvk://")


def test_running_unready_without_ports():
    # reference kubelet.go:1874-1904: RUNNING + unexposed ports => Pending/
    # ContainerCreating
    pod = make_pod(ports=[22])
    info = InstanceInfo(status=PodStatus.RUNNING, ports_exposed=False,
                        requested_ports=["22/tcp"])
    st = translate_status(pod, info, detailed(PodStatus.RUNNING))
    assert st["phase"] == "Pending"
    assert conds(st)["Ready"] == "False"
    assert "waiting" in st["containerStatuses"][0]["state"]


def test_starting():
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.STARTING), None)
    assert st["phase"] == "Pending"
    assert st["containerStatuses"][0]["state"]["waiting"]["reason"] == "ContainerCreating"


def test_exited_success():
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.EXITED),
                          detailed(PodStatus.EXITED, exit_code=0))
    assert st["phase"] == "Succeeded"
    term = st["containerStatuses"][0]["state"]["terminated"]
    assert term["exitCode"] == 0
    assert term["reason"] == "Completed"


def test_exited_failure_nonzero():
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.EXITED),
                          detailed(PodStatus.EXITED, exit_code=2, message="exit code 2"))
    assert st["phase"] == "Failed"
    term = st["containerStatuses"][0]["state"]["terminated"]
    assert term["exitCode"] == 2
    assert term["reason"] == "Error"


def test_exited_message_sniffing():
    # reference kubelet.go:1915-1925: "error"/"fail" in message => Failed even
    # when the completion predicate would pass
    d = detailed(PodStatus.EXITED, exit_code=0, message="completed with error")
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.EXITED), d)
    assert st["phase"] == "Failed"


def test_terminating_keeps_running_phase():
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.TERMINATING),
                          detailed(PodStatus.TERMINATING))
    assert st["phase"] == "Running"
    assert conds(st)["Ready"] == "False"


def test_terminated():
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.TERMINATED),
                          detailed(PodStatus.TERMINATED, exit_code=0))
    assert st["phase"] == "Succeeded"


def test_not_found():
    # reference kubelet.go:1988-2014: NOT_FOUND => Failed/PodDeleted
    st = translate_status(make_pod(), InstanceInfo(status=PodStatus.NOT_FOUND), None)
    assert st["phase"] == "Failed"
    assert st["reason"] == "PodDeleted"


def test_real_start_time_not_fake():
    # deviation from reference quirk kubelet.go:1850 (now - 1h)
    created = time.time() - 5
    info = InstanceInfo(status=PodStatus.RUNNING, ports_exposed=True,
                        creation_time=created)
    st = translate_status(make_pod(), info, detailed(PodStatus.RUNNING))
    from k8s_runpod_kubelet_amd.kube.objects import parse_rfc3339

    assert abs(parse_rfc3339(st["startTime"]) - created) < 2


def test_merge_container_status_preserves_identity():
    # reference mergeContainerStatus (kubelet.go:1798-1820)
    old = [{"name": "main", "containerID": "amdvk://1", "imageID": "img@sha",
            "restartCount": 3, "started": True}]
    new = [{"name": "main", "restartCount": 0, "ready": True}]
    merged = merge_container_status(new, old)
    assert merged[0]["containerID"] == "amdvk://1"
    assert merged[0]["imageID"] == "img@sha"
    assert merged[0]["restartCount"] == 3
    assert merged[0]["started"] is True


def test_qos_class():
    from k8s_runpod_kubelet_amd.provider.status import qos_class

    def pod(resources_list):
        return {"spec": {"containers": [
            {"name": f"c{i}", "resources": r}
            for i, r in enumerate(resources_list)]}}

    assert qos_class(pod([{}])) == "BestEffort"
    assert qos_class(pod([{"requests": {"cpu": "1"}}])) == "Burstable"
    assert qos_class(pod([
        {"limits": {"cpu": "1", "memory": "1Gi"},
         "requests": {"cpu": "1", "memory": "1Gi"}}])) == "Guaranteed"
    # limits-only counts as Guaranteed (requests default to limits)
    assert qos_class(pod([
        {"limits": {"cpu": "1", "memory": "1Gi"}}])) == "Guaranteed"
    # mixed containers -> Burstable
    assert qos_class(pod([
        {"limits": {"cpu": "1", "memory": "1Gi"}},
        {"requests": {"cpu": "1"}}])) == "Burstable"
