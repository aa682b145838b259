"""Runtime (backend) interface — the fakeable seam the reference lacks.

The reference has no mock of its RunPod backend anywhere (its
tests always hit the real cloud API). Here the backend is an interface with
two implementations: the real local ProcessRuntime (process_runtime.py) and an
in-memory FakeRuntime (fake.py) so the entire pod lifecycle runs hermetically
in CPU-only tests.

Surface mirrors the reference Client's backend methods:
- deploy          ← DeployPodREST            (runpod_client.go:522-634)
- terminate       ← TerminatePod             (runpod_client.go:711-739)
- get_status      ← GetPodStatusREST         (runpod_client.go:386-427)
- get_detailed_status ← GetDetailedPodStatus (runpod_client.go:772-818)
- list_instances  ← fetchRunPodInstancesByStatus (kubelet.go:1636-1703)
- healthy         ← the gpuTypes health probe (kubelet.go:320-331)
plus `subscribe` — the event push channel the reference cannot have.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Callable, List, Optional

from .types import DeployParams, DetailedStatus, PodStatus


class Runtime(ABC):
    @abstractmethod
    def deploy(self, params: DeployParams) -> DetailedStatus:
        """Start the pod's containers. Returns the created instance status
        (id + cost). Raises on immediate failure (no GPUs free, bad spec)."""

    @abstractmethod
    def terminate(self, instance_id: str,
                  grace_override_s: float = -1.0) -> None:
        """Request instance stop (SIGTERM, escalating to SIGKILL)."""

    @abstractmethod
    def get_detailed_status(self, instance_id: str) -> DetailedStatus:
        """Full status; NOT_FOUND desired_status for unknown ids (the
        reference maps 404 the same way, runpod_client.go:346-383)."""

    def get_status(self, instance_id: str) -> str:
        return self.get_detailed_status(instance_id).desired_status

    @abstractmethod
    def list_instances(self, statuses: Optional[List[str]] = None) -> List[DetailedStatus]:
        ...

    @abstractmethod
    def get_logs(self, instance_id: str, container: str = "", tail: int = -1) -> str:
        ...

    @abstractmethod
    def healthy(self) -> bool:
        ...

    def subscribe(self, callback: Callable[[str], None]) -> None:
        """Register a push callback fired with an instance_id whenever its
        state changes (ready / exit). Default: no events (poll only)."""

    def remove(self, instance_id: str) -> None:
        """Forget a terminal instance (GC)."""

    def close(self) -> None:
        ...


__all__ = ["Runtime", "DeployParams", "DetailedStatus", "PodStatus"]
