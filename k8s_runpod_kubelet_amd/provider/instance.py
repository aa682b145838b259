"""Provider-side instance tracking (reference InstanceInfo,
pkg/virtual_kubelet/runpod_client.go:87-103)."""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import List, Optional

from ..runtime.types import PodStatus


@dataclass
class InstanceInfo:
    instance_id: str = ""
    status: str = PodStatus.STARTING
    requested_ports: List[str] = field(default_factory=list)
    ports_exposed: bool = False
    creation_time: float = field(default_factory=time.time)
    gpu_indices: List[int] = field(default_factory=list)
    cost_per_hr: float = 0.0
    last_error: str = ""
    ready_time: Optional[float] = None  # first time the pod went Ready (metrics)
    deploying: bool = False  # claim flag: a deploy for this pod is in flight
    # last seen restart/backoff/ephemeral signature (change detection);
    # a tuple since ephemeral-container transitions joined the key
    restart_sig: object = 0
