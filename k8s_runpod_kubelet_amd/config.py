"""Provider configuration.

Counterpart of the reference's ``pkg/config/config.go`` (config.go:8-26) and the
CLI flag surface (cmd/virtual_kubelet/main.go:59-73) — with the reference's dead
knobs actually wired:

- ``max_gpu_price`` is honored by the GPU selector (the reference defines the
  flag at main.go:62 but always passes the constant ``DefaultMaxPrice`` at
  runpod_client.go:1281).
- ``reconcile_interval`` drives the real reconcile ticker (the reference only
  feeds it to informer resync; its loops are hardcoded 10 s/30 s/5 min).
- ``pending_pod_timeout`` replaces the hardcoded 15 min cutoff
  (reference kubelet.go:788); the stuck-terminating ladder thresholds
  (reference kubelet.go:1333/:1285/:1350) are configurable too.
"""

from __future__ import annotations

import dataclasses
import os
from dataclasses import dataclass, field
from typing import List, Optional

import yaml


@dataclass
class Config:
    # Node identity (reference main.go:59-73 flags)
    node_name: str = "virtual-runpod"
    operating_system: str = "Linux"
    internal_ip: str = "127.0.0.1"
    listen_port: int = 10250
    namespace: str = "kube-system"

    # K8s access
    kubeconfig: str = ""

    # Reconcile cadence. The event-driven runtime makes status sync push-based;
    # these are the fallback/periodic cadences (reference: 30 s periodic loop
    # kubelet.go:293, 10 s NotifyPods loop kubelet.go:719, 5 min cleanup
    # kubelet.go:307, 30 s pending retry kubelet.go:735).
    reconcile_interval_s: float = 30.0
    notify_interval_s: float = 10.0
    cleanup_interval_s: float = 300.0
    pending_retry_interval_s: float = 5.0

    # Failure ladders (reference kubelet.go:788, :1333, :1285, :1350)
    pending_pod_timeout_s: float = 900.0
    stuck_reterminate_after_s: float = 300.0
    stuck_statuserr_force_after_s: float = 600.0
    stuck_force_after_s: float = 900.0

    # GPU policy. max_gpu_price is kept for Helm-values compatibility and
    # remapped to a max fractional "cost" of a GPU (its current busy/HBM
    # occupancy score in [0,1]); selector skips GPUs costlier than this.
    max_gpu_price: float = 0.5
    gpu_memory_default_gb: int = 16  # reference runpod_client.go:1189 default
    datacenter_ids: List[str] = field(default_factory=list)

    # spec.restartPolicy default for pods that arrive without one (a real
    # apiserver defaults it to Always at admission — k8s semantics). Set to
    # "Never" for the reference's run-to-completion model.
    restart_policy_default: str = "Always"

    # Ops
    health_server_address: str = ":8080"
    # Bearer token for the state-mutating admin endpoints (cordon/uncordon)
    # when called from non-loopback peers; empty = loopback-only. Env
    # AMDVK_ADMIN_TOKEN overrides when this is unset.
    admin_token: str = ""
    # Bearer token for the kubelet API (:10250 — logs/exec/stats) from
    # non-loopback peers; empty = loopback-only (the real kubelet
    # authenticates this surface via webhook/x509). Env
    # AMDVK_KUBELET_TOKEN overrides when unset.
    kubelet_api_token: str = ""
    log_level: str = "info"
    heartbeat_interval_s: float = 300.0  # 0 disables (reference kubelet.go:73)
    registration_endpoint: str = ""  # optional registration hook, off by default

    # Local node backend (no cloud)
    state_dir: str = "/var/lib/amd-virtual-kubelet"
    sysfs_root: str = "/sys"
    cgroup_root: str = "/sys/fs/cgroup"
    cgroup_parent: str = "amdvk.slice"
    # pod PID/UTS namespace isolation (auto-degrades without CAP_SYS_ADMIN)
    pod_namespaces: bool = True
    runtime: str = "process"  # process | fake
    pod_log_dir: str = ""  # defaults to <state_dir>/logs
    # per-container log cap (copytruncate rotation into the --previous
    # slot; kubelet default is 10 MB — we default higher for GPU jobs)
    pod_log_max_bytes: int = 50 * 1024 * 1024
    # OCI image execution: pods whose image is present in the local store
    # run inside it (mount-ns overlay + pivot_root, or chroot fallback).
    # The store holds OCI image layouts/archives imported via
    # `python -m k8s_runpod_kubelet_amd.runtime.imagetool`.
    image_store_dir: str = ""       # defaults to <state_dir>/images
    image_isolation: str = "auto"   # auto | mountns | chroot
    # host paths bound read-only into GPU image pods (driver userspace —
    # the thin-image + host-ROCm pattern); "src[:dst[:ro|rw]]"
    image_gpu_binds: List[str] = field(default_factory=lambda: ["/opt/rocm"])
    image_extra_binds: List[str] = field(default_factory=list)
    # optional in-cluster registry: images missing from the local store are
    # pulled from here at deploy time (base URL, e.g. http://registry:5000;
    # empty = no in-kubelet pulls — operators feed the store via imagetool)
    image_registry: str = ""
    image_registry_token: str = ""  # or env AMDVK_REGISTRY_TOKEN
    pod_controller_workers: int = 4  # reference uses 1 (main.go:263)

    # Node-pressure eviction (kubelet memory.available hard-eviction
    # analogue; 0 disables — this node may share memory with non-pod
    # workloads, so the operator opts in with a threshold, e.g. 512 for
    # the kubelet's classic memory.available<...Mi signal)
    eviction_memory_threshold_mb: int = 0
    eviction_interval_s: float = 10.0
    # fail pods bound to a GPU that goes unhealthy (RAS uncorrectable)
    # mid-run, so controllers reschedule them instead of wedging on a
    # dead device; the GPU itself is cordoned by the health re-probe
    evict_on_gpu_failure: bool = True

    # GPU inventory overrides (mostly for tests / CPU-only dev)
    gpu_count_override: int = -1
    gpu_vram_gb_override: int = -1

    def resolved_pod_log_dir(self) -> str:
        return self.pod_log_dir or os.path.join(self.state_dir, "logs")

    def resolved_image_store_dir(self) -> str:
        return self.image_store_dir or os.path.join(self.state_dir, "images")


def load_config(path: Optional[str]) -> Config:
    """Load YAML config overlaid on defaults (reference main.go:76-90).

    Unknown keys are rejected loudly instead of silently ignored.
    """
    cfg = Config()
    if not path:
        return cfg
    with open(path, "r", encoding="utf-8") as fh:
        data = yaml.safe_load(fh) or {}
    if not isinstance(data, dict):
        raise ValueError(f"provider config {path!r}: expected a mapping")
    names = {f.name for f in dataclasses.fields(Config)}
    for key, value in data.items():
        norm = key.replace("-", "_")
        if norm not in names:
            raise ValueError(f"provider config {path!r}: unknown key {key!r}")
        setattr(cfg, norm, value)
    if cfg.datacenter_ids is None:
        cfg.datacenter_ids = []
    if isinstance(cfg.datacenter_ids, str):
        cfg.datacenter_ids = [s.strip() for s in cfg.datacenter_ids.split(",") if s.strip()]
    return cfg
