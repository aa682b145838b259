"""FakeRuntime — in-memory backend for hermetic lifecycle tests.

Fills the reference's biggest testing gap (no mock of the RunPod
API exists; its tests require real cloud credentials). Supports scripted
transition delays and fault injection so the provider's retry ladders and
failure paths are testable without hardware.
"""

from __future__ import annotations

import secrets
import threading
import time
from typing import Callable, Dict, List, Optional

from .base import Runtime
from .types import (
    ContainerRuntimeInfo,
    DeployParams,
    DetailedStatus,
    Instance,
    PodStatus,
)


class FakeRuntime(Runtime):
    def __init__(
        self,
        gpu_count: int = 8,
        start_delay_s: float = 0.0,
        deploy_error: Optional[str] = None,
        healthy: bool = True,
    ):
        self.gpu_count = gpu_count
        self.start_delay_s = start_delay_s
        self.deploy_error = deploy_error  # if set, deploy() raises with this
        self._healthy = healthy
        self._lock = threading.RLock()
        self.instances: Dict[str, Instance] = {}
        self._subscribers: List[Callable[[str], None]] = []
        self._free_gpus = list(range(gpu_count))
        self.deploy_calls = 0
        self.terminate_calls = 0

    # -- test controls --

    def set_healthy(self, value: bool) -> None:
        self._healthy = value

    def fail_instance(self, instance_id: str, exit_code: int = 1, message: str = "") -> None:
        with self._lock:
            inst = self.instances[instance_id]
            for c in inst.containers:
                c.exit_code = exit_code
                c.finished_at = time.time()
                c.ready = False
                c.message = message
            inst.desired_status = PodStatus.EXITED
            self._release(inst)
        self._notify(instance_id)

    def complete_instance(self, instance_id: str, exit_code: int = 0) -> None:
        self.fail_instance(instance_id, exit_code=exit_code, message="completed")

    def vanish_instance(self, instance_id: str) -> None:
        """Instance disappears from the backend (reference NOT_FOUND path,
        kubelet.go:1708 handleMissingRunPodInstance)."""
        with self._lock:
            inst = self.instances.pop(instance_id, None)
            if inst is not None:
                self._release(inst)

    # -- Runtime interface --

    def deploy(self, params: DeployParams) -> DetailedStatus:
        with self._lock:
            self.deploy_calls += 1
            if self.deploy_error:
                raise RuntimeError(self.deploy_error)
            if params.gpu_count > len(self._free_gpus):
                from ..gpu.binder import PlacementError

                raise PlacementError(
                    f"need {params.gpu_count} GPUs, only {len(self._free_gpus)} free"
                )
            gpus = [self._free_gpus.pop(0) for _ in range(params.gpu_count)]
            inst = Instance(
                id="fake-" + secrets.token_hex(4),
                pod_key=params.pod_key,
                params=params,
                gpu_indices=gpus,
                cost_per_hr=0.1 * max(1, params.gpu_count),
            )
            inst.containers = [
                ContainerRuntimeInfo(name=c.name, pid=1000 + i, started_at=time.time())
                for i, c in enumerate(params.containers)
            ] or [ContainerRuntimeInfo(name="main", pid=1000, started_at=time.time())]
            self.instances[inst.id] = inst

        if self.start_delay_s > 0:
            timer = threading.Timer(self.start_delay_s, self._go_running, args=(inst.id,))
            timer.daemon = True
            timer.start()
        else:
            self._go_running(inst.id)
        return self.get_detailed_status(inst.id)

    def _go_running(self, instance_id: str) -> None:
        with self._lock:
            inst = self.instances.get(instance_id)
            if inst is None or inst.desired_status != PodStatus.STARTING:
                return
            inst.desired_status = PodStatus.RUNNING
            for c in inst.containers:
                c.ready = True
        self._notify(instance_id)

    def terminate(self, instance_id: str,
                  grace_override_s: float = -1.0) -> None:
        with self._lock:
            self.terminate_calls += 1
            inst = self.instances.get(instance_id)
            if inst is None:
                return
            if inst.desired_status in (PodStatus.EXITED, PodStatus.TERMINATED):
                inst.desired_status = PodStatus.TERMINATED
            else:
                inst.desired_status = PodStatus.TERMINATED
                for c in inst.containers:
                    if c.exit_code is None:
                        c.exit_code = 0
                        c.finished_at = time.time()
                        c.ready = False
            self._release(inst)
        self._notify(instance_id)

    def _release(self, inst: Instance) -> None:
        for g in inst.gpu_indices:
            if g not in self._free_gpus:
                self._free_gpus.append(g)
        inst.gpu_indices = list(inst.gpu_indices)
        self._free_gpus.sort()

    def get_detailed_status(self, instance_id: str) -> DetailedStatus:
        with self._lock:
            inst = self.instances.get(instance_id)
            if inst is None:
                return DetailedStatus(id=instance_id, desired_status=PodStatus.NOT_FOUND)
            ports: Dict[int, int] = {}
            if inst.desired_status == PodStatus.RUNNING:
                for c in inst.params.containers:
                    for p in c.tcp_ports:
                        ports[p] = p
            return DetailedStatus(
                id=inst.id,
                desired_status=inst.desired_status,
                port_mappings=ports,
                containers=[ContainerRuntimeInfo(**vars(c)) for c in inst.containers],
                gpu_indices=list(inst.gpu_indices),
                cost_per_hr=inst.cost_per_hr,
                created_at=inst.created_at,
            )

    def list_instances(self, statuses: Optional[List[str]] = None) -> List[DetailedStatus]:
        with self._lock:
            ids = list(self.instances)
        out = [self.get_detailed_status(i) for i in ids]
        if statuses:
            out = [s for s in out if s.desired_status in statuses]
        return out

    def instance_for_pod(self, pod_key: str) -> Optional[str]:
        with self._lock:
            for inst in self.instances.values():
                if inst.pod_key == pod_key and inst.desired_status != PodStatus.TERMINATED:
                    return inst.id
        return None

    def get_logs(self, instance_id: str, container: str = "", tail: int = -1) -> str:
        return f"fake logs for {instance_id}\n"

    def healthy(self) -> bool:
        return self._healthy

    def subscribe(self, callback: Callable[[str], None]) -> None:
        self._subscribers.append(callback)

    def remove(self, instance_id: str) -> None:
        with self._lock:
            inst = self.instances.pop(instance_id, None)
            if inst is not None:
                self._release(inst)

    def _notify(self, instance_id: str) -> None:
        for cb in list(self._subscribers):
            try:
                cb(instance_id)
            except Exception:
                pass
