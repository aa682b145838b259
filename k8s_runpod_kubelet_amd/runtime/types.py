"""Backend wire types — local equivalents of the reference's client types
(reference pkg/virtual_kubelet/runpod_client.go:55-140: PodStatus enum,
RunPodInstance, InstanceInfo, DetailedStatus/RuntimeInfo/MachineInfo).

The status vocabulary is kept verbatim so the provider's status translation
(provider/status.py) matches the reference's semantics
(kubelet.go:1848-2024) state for state.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .probes import ProbeSpec  # noqa: F401 (re-exported for ContainerSpec)


class PodStatus:
    RUNNING = "RUNNING"
    STARTING = "STARTING"
    TERMINATING = "TERMINATING"
    TERMINATED = "TERMINATED"
    NOT_FOUND = "NOT_FOUND"
    EXITED = "EXITED"


@dataclass
class ContainerRuntimeInfo:
    """Per-container runtime record (reference RuntimeInfo is single-container;
    we track all containers — SURVEY §7.4 parity-plus)."""
    name: str
    pid: int = -1
    started_at: float = 0.0
    finished_at: float = 0.0
    exit_code: Optional[int] = None
    ready: bool = False
    message: str = ""
    # restartPolicy bookkeeping: completed restarts; while a restart is
    # pending, the wall-clock deadline of its backoff (CrashLoopBackOff
    # state); consecutive-crash streak driving the exponential delay
    restart_count: int = 0
    backoff_until: float = 0.0
    crash_streak: int = 0
    # resolved image identity for status.imageID (ref@manifest-digest),
    # empty for host-binary containers
    image_id: str = ""
    # in-kubelet registry pull time for this launch (0 = local cache hit)
    pull_seconds: float = 0.0


@dataclass
class DetailedStatus:
    """Reference DetailedStatus analogue (runpod_client.go:105-140)."""
    id: str
    desired_status: str = PodStatus.NOT_FOUND
    # "exposed" TCP ports actually listening — the portMappings analogue
    # (machine.portMappings in the reference, kubelet.go:566-605 gate).
    port_mappings: Dict[int, int] = field(default_factory=dict)
    # pod identity (for orphan GC after kubelet restarts)
    namespace: str = ""
    name: str = ""
    containers: List[ContainerRuntimeInfo] = field(default_factory=list)
    init_containers: List[ContainerRuntimeInfo] = field(default_factory=list)
    ephemeral_containers: List[ContainerRuntimeInfo] = field(
        default_factory=list)
    gpu_indices: List[int] = field(default_factory=list)
    cost_per_hr: float = 0.0
    last_error: str = ""
    created_at: float = 0.0

    @property
    def exit_code(self) -> Optional[int]:
        """Aggregate exit code: a failed init container's code, else first
        nonzero main code, else 0 once all mains finished."""
        for c in self.init_containers:
            if c.exit_code:
                return c.exit_code
        codes = [c.exit_code for c in self.containers]
        if any(c is None for c in codes) or not codes:
            return None
        for code in codes:
            if code:
                return code
        return 0

    @property
    def completion_message(self) -> str:
        for c in self.init_containers:
            if c.exit_code:
                return (c.message
                        or f"init container {c.name} exited with code {c.exit_code}")
        for c in self.containers:
            if c.message:
                return c.message
        for c in self.containers:
            if c.exit_code:
                return f"container {c.name} exited with code {c.exit_code}"
        return "completed"


def is_successful_completion(status: DetailedStatus) -> bool:
    """Reference IsSuccessfulCompletion (runpod_client.go:820-843): no runtime
    info => not successful; exit code 0 => success; else message sniffing."""
    code = status.exit_code
    if code is None:
        return False
    if code == 0:
        return True
    msg = status.completion_message.lower()
    return "success" in msg or "completed" in msg


@dataclass
class VolumeMount:
    """containers[].volumeMounts entry."""
    name: str
    mount_path: str
    read_only: bool = False
    sub_path: str = ""


@dataclass
class VolumeSource:
    """spec.volumes entry, resolved at translation time.

    kind: "emptyDir" (per-pod scratch dir, shared between the pod's
    containers, survives container restarts — k8s semantics; medium=Memory
    mounts a tmpfs sized by sizeLimit in mount-ns mode, plain dir in the
    chroot fallback), "hostPath" (host
    directory/file bound in; mount-namespace isolation mode only), or
    "files" (secret/configMap projected to files — content fetched from
    the API at translation time like env extraction)."""
    kind: str
    host_path: str = ""
    files: Dict[str, str] = field(default_factory=dict)
    file_mode: int = 0o644
    # emptyDir.medium ("Memory" -> tmpfs in mountns mode; plain dir in
    # chroot fallback) and emptyDir.sizeLimit (tmpfs size= cap)
    medium: str = ""
    size_limit_bytes: int = 0


@dataclass
class ContainerSpec:
    name: str
    image: str = ""
    command: List[str] = field(default_factory=list)
    args: List[str] = field(default_factory=list)
    env: Dict[str, str] = field(default_factory=dict)
    working_dir: str = ""
    tcp_ports: List[int] = field(default_factory=list)
    volume_mounts: List[VolumeMount] = field(default_factory=list)
    # securityContext.runAsUser/runAsGroup (container overrides pod; -1 =
    # inherit the kubelet's credentials)
    run_as_uid: int = -1
    run_as_gid: int = -1
    # livenessProbe / readinessProbe / startupProbe (None = absent)
    liveness: Optional["ProbeSpec"] = None
    readiness: Optional["ProbeSpec"] = None
    # startupProbe gates the other two (k8s: the container is not Started
    # until it passes; failureThreshold exhausted kills the container)
    startup: Optional["ProbeSpec"] = None
    # spec.containers[].terminationMessagePath: file the container writes
    # its exit message to, surfaced in terminated status (image pods)
    termination_message_path: str = "/dev/termination-log"
    # terminationMessagePolicy: File | FallbackToLogsOnError (tail of the
    # container log used when the file is empty and the container failed)
    termination_message_policy: str = "File"
    # imagePullPolicy: IfNotPresent | Always | Never ("" = k8s default:
    # Always for :latest/untagged refs, IfNotPresent otherwise)
    image_pull_policy: str = ""
    # securityContext.runAsNonRoot: refuse to start if the effective uid
    # resolves to root (CreateContainerConfigError analogue)
    run_as_non_root: bool = False
    # securityContext.readOnlyRootFilesystem: rootfs mounted read-only
    # (volume mounts stay writable); mountns/overlay mode only
    read_only_root_fs: bool = False
    # lifecycle hooks (exec/httpGet/sleep handlers): postStart runs right
    # after the container starts (failure kills it into restartPolicy);
    # preStop runs before SIGTERM, inside the grace window
    post_start: Optional["ProbeSpec"] = None
    pre_stop: Optional["ProbeSpec"] = None


@dataclass
class DeployParams:
    """Local-deploy parameter set — the assembled output of spec translation
    (reference PrepareRunPodParameters, runpod_client.go:1248-1377)."""
    pod_key: str  # "<namespace>-<name>" (reference map key convention)
    name: str
    namespace: str = "default"
    containers: List[ContainerSpec] = field(default_factory=list)
    # spec.dnsConfig rendered to a resolv.conf ("" = node default; image
    # pods get it written into /etc/resolv.conf like the kubelet)
    resolv_conf: str = ""
    # spec.initContainers: run sequentially to completion before the main
    # containers start; any nonzero exit fails the pod
    init_containers: List[ContainerSpec] = field(default_factory=list)
    # spec.restartPolicy: Never | OnFailure | Always (k8s default Always)
    restart_policy: str = "Never"
    # spec.terminationGracePeriodSeconds: SIGTERM → SIGKILL ladder window
    termination_grace_s: float = 10.0
    # spec.activeDeadlineSeconds: pod killed + Failed/DeadlineExceeded after
    # this long (0 = unlimited)
    active_deadline_s: float = 0.0
    env: Dict[str, str] = field(default_factory=dict)  # pod-level (merged into all)
    gpu_count: int = 0
    gpu_memory_bytes: int = 0  # total across the GPU set
    max_gpu_cost: float = 0.5
    requested_ports: List[str] = field(default_factory=list)  # "8080/tcp" style
    cloud_type: str = "SECURE"
    datacenter_ids: List[str] = field(default_factory=list)
    template_id: str = ""
    registry_auth_id: str = ""
    cpu_limit: str = ""      # cgroup cpu.max, e.g. "200000 100000"
    memory_limit: str = ""   # cgroup memory.max bytes or "max"
    # spec.volumes resolved (see VolumeSource); consumed by image-backed
    # containers (host-process pods already see the host filesystem)
    volumes: Dict[str, VolumeSource] = field(default_factory=dict)
    labels: Dict[str, str] = field(default_factory=dict)
    # k8s pod-hostname semantics (spec.hostname, else pod name); applied in
    # the pod's own UTS namespace when namespace isolation is available
    hostname: str = ""
    # securityContext.fsGroup: emptyDir/projected volumes are group-owned
    # and group-writable by this gid (-1 = unset)
    fs_group: int = -1
    # spec.hostAliases -> extra /etc/hosts lines in image pods
    host_aliases: List[tuple] = field(default_factory=list)  # (ip, [names])


@dataclass
class Instance:
    """A deployed local instance (reference RunPodInstance analogue)."""
    id: str
    pod_key: str
    params: DeployParams
    gpu_indices: List[int] = field(default_factory=list)
    desired_status: str = PodStatus.STARTING
    containers: List[ContainerRuntimeInfo] = field(default_factory=list)
    init_containers: List[ContainerRuntimeInfo] = field(default_factory=list)
    # kubectl-debug ephemeral containers: run alongside, never restarted,
    # never gate readiness or completion
    ephemeral_containers: List[ContainerRuntimeInfo] = field(
        default_factory=list)
    ephemeral_specs: List["ContainerSpec"] = field(default_factory=list)
    init_index: int = 0  # next init container to run
    cgroup_dir: str = ""
    created_at: float = field(default_factory=time.time)
    cost_per_hr: float = 0.0
    last_error: str = ""
    deadline_exceeded: bool = False  # activeDeadlineSeconds fired
    # image-backed execution mode: "" (host binaries) | "mountns" | "chroot"
    # — drives how kubectl-exec/probes enter the container (setns vs chroot)
    image_mode: str = ""
