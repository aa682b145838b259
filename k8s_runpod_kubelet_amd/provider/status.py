"""Backend→Kubernetes pod status translation.

Faithful to the reference's translateRunPodStatus
(pkg/virtual_kubelet/kubelet.go:1848-2024) state for state:

- RUNNING: phase Running; Ready iff requested ports are exposed, else the
  container shows ContainerCreating and the pod stays unready,
- STARTING: phase Pending / ContainerCreating,
- EXITED: message-sniffed success → Succeeded/Completed, else Failed/Error
  with exit code (kubelet.go:1906-1954),
- TERMINATING: phase Running (kubelet.go:1956),
- TERMINATED: phase Succeeded (kubelet.go:1972),
- NOT_FOUND: phase Failed, reason PodDeleted (kubelet.go:1988),
- four conditions PodScheduled/Initialized/Ready/ContainersReady.

Deliberate deviations from reference quirks (documented, not replicated):
- real start/ready times instead of the fake "now − 1 h" (kubelet.go:1850),
- hostIP/podIP are the node's internal IP (host-process pods) instead of the
  placeholders 10.0.0.1/10.0.0.2 (kubelet.go:2016-2017).
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..kube.objects import now_rfc3339, ts_rfc3339
from ..runtime.types import DetailedStatus, PodStatus, is_successful_completion
from .instance import InstanceInfo

PHASE_MAP = {
    # reference translateRunPodStatusToPhase (kubelet.go:978-995)
    PodStatus.RUNNING: "Running",
    PodStatus.STARTING: "Pending",
    PodStatus.EXITED: "Succeeded",
    PodStatus.TERMINATING: "Running",
    PodStatus.TERMINATED: "Succeeded",
    PodStatus.NOT_FOUND: "Unknown",
}


def translate_status_to_phase(status: str) -> str:
    return PHASE_MAP.get(status, "Unknown")


def _conditions(scheduled: bool, initialized: bool, ready: bool,
                reason: str = "", message: str = "") -> List[Dict[str, Any]]:
    def cond(ctype: str, ok: bool) -> Dict[str, Any]:
        c = {
            "type": ctype,
            "status": "True" if ok else "False",
            "lastTransitionTime": now_rfc3339(),
        }
        if not ok and reason:
            c["reason"] = reason
        if not ok and message:
            c["message"] = message
        return c

    return [
        cond("PodScheduled", scheduled),
        cond("Initialized", initialized),
        cond("Ready", ready),
        cond("ContainersReady", ready),
    ]


def _container_statuses(
    pod: Dict[str, Any],
    detailed: Optional[DetailedStatus],
    state: str,
    ready: bool,
    exit_code: int = 0,
    reason: str = "",
    message: str = "",
) -> List[Dict[str, Any]]:
    out = []
    runtime_by_name = {}
    if detailed is not None:
        runtime_by_name = {c.name: c for c in detailed.containers}
    for container in pod.get("spec", {}).get("containers", []) or []:
        name = container.get("name", "")
        rt = runtime_by_name.get(name)
        started_at = ts_rfc3339(rt.started_at) if rt and rt.started_at else now_rfc3339()
        cs: Dict[str, Any] = {
            "name": name,
            "image": container.get("image", ""),
            "imageID": rt.image_id if rt else "",
            "restartCount": rt.restart_count if rt else 0,
            "ready": ready,
        }
        if rt and rt.pid > 0:
            cs["containerID"] = f"amdvk://{rt.pid}"
        if state == "running":
            # Per-container fidelity under restartPolicy: a crashed
            # container awaiting its backoff shows CrashLoopBackOff; one
            # that exited without restart shows terminated.
            if rt is not None and rt.exit_code is not None:
                if rt.backoff_until:
                    cs["ready"] = False
                    cs["state"] = {"waiting": {
                        "reason": "CrashLoopBackOff",
                        "message": rt.message or
                        f"back-off restarting failed container (exit "
                        f"{rt.exit_code})",
                    }}
                else:
                    cs["ready"] = False
                    cs["state"] = {"terminated": {
                        "exitCode": rt.exit_code,
                        "reason": "Completed" if rt.exit_code == 0 else "Error",
                        "message": rt.message,
                        "startedAt": started_at,
                        "finishedAt": ts_rfc3339(rt.finished_at)
                        if rt.finished_at else now_rfc3339(),
                    }}
                out.append(cs)
                continue
            cs["state"] = {"running": {"startedAt": started_at}}
            cs["started"] = True
        elif state == "waiting":
            cs["state"] = {"waiting": {"reason": reason or "ContainerCreating"}}
        else:  # terminated
            code = rt.exit_code if rt and rt.exit_code is not None else exit_code
            finished = ts_rfc3339(rt.finished_at) if rt and rt.finished_at else now_rfc3339()
            cs["state"] = {
                "terminated": {
                    "exitCode": code,
                    "reason": reason or ("Completed" if code == 0 else "Error"),
                    "message": message,
                    "startedAt": started_at,
                    "finishedAt": finished,
                }
            }
        out.append(cs)
    return out


def qos_class(pod: Dict[str, Any]) -> str:
    """k8s QoS classification (kubectl shows status.qosClass): Guaranteed
    when every container has requests==limits for both cpu and memory;
    BestEffort when no container has any request/limit; else Burstable."""
    containers = pod.get("spec", {}).get("containers", []) or []
    any_set = False
    guaranteed = bool(containers)
    for c in containers:
        res = c.get("resources", {}) or {}
        req = res.get("requests", {}) or {}
        lim = res.get("limits", {}) or {}
        if req or lim:
            any_set = True
        for key in ("cpu", "memory"):
            limv = lim.get(key)
            if limv is None or req.get(key, limv) != limv:
                guaranteed = False
    if guaranteed and any_set:
        return "Guaranteed"
    return "Burstable" if any_set else "BestEffort"


def _ephemeral_container_statuses(
    detailed: Optional[DetailedStatus],
) -> List[Dict[str, Any]]:
    """spec.ephemeralContainers status projection (kubectl debug)."""
    out = []
    if detailed is None:
        return out
    for c in getattr(detailed, "ephemeral_containers", []):
        cs: Dict[str, Any] = {"name": c.name, "image": "", "imageID": "",
                              "restartCount": 0, "ready": False}
        if c.pid > 0:
            cs["containerID"] = f"amdvk://{c.pid}"
        if c.exit_code is None:
            cs["state"] = {"running": {
                "startedAt": ts_rfc3339(c.started_at) if c.started_at
                else now_rfc3339()}}
        else:
            cs["state"] = {"terminated": {
                "exitCode": c.exit_code,
                "reason": "Completed" if c.exit_code == 0 else "Error",
                "startedAt": ts_rfc3339(c.started_at) if c.started_at
                else now_rfc3339(),
                "finishedAt": ts_rfc3339(c.finished_at) if c.finished_at
                else now_rfc3339(),
            }}
        out.append(cs)
    return out


def _init_container_statuses(
    detailed: Optional[DetailedStatus],
) -> List[Dict[str, Any]]:
    """spec.initContainers status projection (running / terminated)."""
    out = []
    if detailed is None:
        return out
    for c in detailed.init_containers:
        cs: Dict[str, Any] = {
            "name": c.name, "image": "", "imageID": "",
            "restartCount": 0, "ready": c.exit_code == 0,
        }
        if c.pid > 0:
            cs["containerID"] = f"amdvk://{c.pid}"
        if c.exit_code is None:
            cs["state"] = {"running": {
                "startedAt": ts_rfc3339(c.started_at) if c.started_at
                else now_rfc3339()}}
        else:
            cs["state"] = {"terminated": {
                "exitCode": c.exit_code,
                "reason": "Completed" if c.exit_code == 0 else "Error",
                "message": c.message,
                "startedAt": ts_rfc3339(c.started_at) if c.started_at
                else now_rfc3339(),
                "finishedAt": ts_rfc3339(c.finished_at) if c.finished_at
                else now_rfc3339(),
            }}
        out.append(cs)
    return out


def translate_status(
    pod: Dict[str, Any],
    info: InstanceInfo,
    detailed: Optional[DetailedStatus],
    node_ip: str = "127.0.0.1",
) -> Dict[str, Any]:
    status = info.status
    start_time = ts_rfc3339(info.creation_time)
    base: Dict[str, Any] = {
        "hostIP": node_ip,
        "podIP": node_ip,
        "startTime": start_time,
        "qosClass": qos_class(pod),
    }
    inits = _init_container_statuses(detailed)
    if inits:
        base["initContainerStatuses"] = inits
    ephs = _ephemeral_container_statuses(detailed)
    if ephs:
        base["ephemeralContainerStatuses"] = ephs
    if status == PodStatus.RUNNING:
        ready = info.ports_exposed
        if ready:
            base.update(
                phase="Running",
                conditions=_conditions(True, True, True),
                containerStatuses=_container_statuses(pod, detailed, "running", True),
            )
        else:
            # RUNNING but ports not yet exposed: Pending/ContainerCreating
            # (reference kubelet.go:1874-1904)
            base.update(
                phase="Pending",
                conditions=_conditions(True, True, False, "ContainersNotReady",
                                       "waiting for ports to be exposed"),
                containerStatuses=_container_statuses(
                    pod, detailed, "waiting", False, reason="ContainerCreating"
                ),
            )
    elif status == PodStatus.STARTING:
        base.update(
            phase="Pending",
            conditions=_conditions(True, True, False, "ContainersNotReady",
                                   "containers are starting"),
            containerStatuses=_container_statuses(
                pod, detailed, "waiting", False, reason="ContainerCreating"
            ),
        )
    elif status == PodStatus.EXITED:
        success = detailed is not None and is_successful_completion(detailed)
        message = detailed.completion_message if detailed is not None else ""
        # Message sniffing parity (kubelet.go:1915-1925): "error"/"fail" in
        # the completion message forces failure.
        low = message.lower()
        if "error" in low or "fail" in low:
            success = False
        code = detailed.exit_code if detailed is not None and detailed.exit_code is not None else (0 if success else 1)
        if success:
            base.update(
                phase="Succeeded",
                conditions=_conditions(True, True, False, "PodCompleted", message),
                containerStatuses=_container_statuses(
                    pod, detailed, "terminated", False, exit_code=code,
                    reason="Completed", message=message,
                ),
            )
        else:
            base.update(
                phase="Failed",
                conditions=_conditions(True, True, False, "PodFailed", message),
                containerStatuses=_container_statuses(
                    pod, detailed, "terminated", False,
                    exit_code=code if code else 1, reason="Error", message=message,
                ),
            )
            if detailed is not None and detailed.last_error == "DeadlineExceeded":
                base["reason"] = "DeadlineExceeded"  # activeDeadlineSeconds
    elif status == PodStatus.TERMINATING:
        base.update(
            phase="Running",
            conditions=_conditions(True, True, False, "Terminating", "pod is terminating"),
            containerStatuses=_container_statuses(pod, detailed, "running", False),
        )
    elif status == PodStatus.TERMINATED:
        base.update(
            phase="Succeeded",
            conditions=_conditions(True, True, False, "PodTerminated", ""),
            containerStatuses=_container_statuses(
                pod, detailed, "terminated", False, reason="Completed"
            ),
        )
    else:  # NOT_FOUND
        base.update(
            phase="Failed",
            reason="PodDeleted",
            message="backend instance no longer exists",
            conditions=_conditions(True, True, False, "PodDeleted",
                                   "backend instance no longer exists"),
            containerStatuses=_container_statuses(
                pod, detailed, "terminated", False, exit_code=1, reason="Error",
                message="instance not found",
            ),
        )
    return base


def merge_container_status(
    new_statuses: List[Dict[str, Any]], old_statuses: List[Dict[str, Any]]
) -> List[Dict[str, Any]]:
    """Preserve ContainerID/ImageID/Started/RestartCount across rewrites
    (reference mergeContainerStatus, kubelet.go:1798-1820)."""
    old_by_name = {c.get("name"): c for c in old_statuses}
    for cs in new_statuses:
        old = old_by_name.get(cs.get("name"))
        if not old:
            continue
        if not cs.get("containerID") and old.get("containerID"):
            cs["containerID"] = old["containerID"]
        if not cs.get("imageID") and old.get("imageID"):
            cs["imageID"] = old["imageID"]
        if "started" not in cs and "started" in old:
            cs["started"] = old["started"]
        # Monotonic: the runtime's live count wins once restarts happen.
        cs["restartCount"] = max(old.get("restartCount", 0),
                                 cs.get("restartCount", 0))
    return new_statuses
