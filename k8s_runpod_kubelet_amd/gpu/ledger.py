"""Per-GPU HBM ledger — the ``runpodAvailable`` analogue made per-GPU.

The reference keeps a single backend-availability boolean
(reference pkg/virtual_kubelet/kubelet.go:320-331) and no notion of capacity
beyond the cloud's. Here every GPU has its own schedulability (health from RAS
counters/thermals) and an HBM reservation ledger so the binder can bin-pack
pods by headroom. Assignments are also persisted into pod annotations
(``amd.com/gpu-ids``) so the ledger is reconstructible after a kubelet
restart (reference analogue: `runpod.io/pod-id` annotation + LoadRunning,
kubelet.go:1380).
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .inventory import Gpu, Inventory


@dataclass
class Reservation:
    pod_key: str
    gpu_indices: List[int]
    bytes_per_gpu: int


@dataclass
class GpuState:
    gpu: Gpu
    reserved_bytes: int = 0
    pod_keys: List[str] = field(default_factory=list)
    cordoned: bool = False  # operator/admin gate, separate from health
    # monotonic timestamp of the last release on this GPU. A freshly freed
    # GPU is "settling": the exited pod's KFD context teardown (VRAM unmap,
    # queue destruction) runs asynchronously in the kernel for a few hundred
    # ms and serializes against the next pod's HIP init — measured on MI355X
    # as ~300 ms init back-to-back vs ~160 ms after a 1 s gap
    # (profiles/pw_timing.txt). The binder prefers settled GPUs.
    last_freed_at: float = 0.0

    @property
    def schedulable(self) -> bool:
        return self.gpu.healthy and not self.cordoned

    @property
    def occupied(self) -> bool:
        """``amd.com/gpu`` is an exclusive countable resource (Kubernetes
        device-plugin semantics): one pod's GPU claim owns the whole GPU."""
        return bool(self.pod_keys)

    @property
    def headroom_bytes(self) -> int:
        """Free HBM available for new reservations: total minus the larger of
        (ledger reservations, live measured use)."""
        used = max(self.reserved_bytes, self.gpu.vram_used_bytes)
        return max(0, self.gpu.vram_total_bytes - used)

    def cost(self) -> float:
        """Occupancy score in [0,1] — the max_gpu_price remap: a GPU's
        "price" is how occupied it already is (HBM reservations + busy%)."""
        mem = self.reserved_bytes / self.gpu.vram_total_bytes if self.gpu.vram_total_bytes else 1.0
        busy = max(0, self.gpu.busy_percent) / 100.0
        return min(1.0, 0.5 * mem + 0.5 * busy)


class Ledger:
    def __init__(self, inventory: Inventory):
        self.inventory = inventory
        self._lock = threading.RLock()
        self.states: Dict[int, GpuState] = {}
        self.reservations: Dict[str, Reservation] = {}

    def sync_inventory(self) -> None:
        with self._lock:
            for gpu in self.inventory.gpus:
                if gpu.index in self.states:
                    self.states[gpu.index].gpu = gpu
                else:
                    self.states[gpu.index] = GpuState(gpu=gpu)
            live = {g.index for g in self.inventory.gpus}
            for idx in list(self.states):
                if idx not in live:
                    del self.states[idx]

    def reserve(self, pod_key: str, gpu_indices: List[int], bytes_per_gpu: int) -> None:
        with self._lock:
            if pod_key in self.reservations:
                raise ValueError(f"pod {pod_key} already holds a GPU reservation")
            for idx in gpu_indices:
                state = self.states.get(idx)
                if state is None:
                    raise ValueError(f"unknown GPU index {idx}")
                if not state.schedulable:
                    raise ValueError(f"GPU {idx} is not schedulable")
                if state.occupied:
                    raise ValueError(f"GPU {idx} is already bound to {state.pod_keys}")
            for idx in gpu_indices:
                state = self.states[idx]
                state.reserved_bytes += bytes_per_gpu
                state.pod_keys.append(pod_key)
            self.reservations[pod_key] = Reservation(pod_key, list(gpu_indices), bytes_per_gpu)

    def release(self, pod_key: str) -> Optional[Reservation]:
        with self._lock:
            res = self.reservations.pop(pod_key, None)
            if res is None:
                return None
            now = time.monotonic()
            for idx in res.gpu_indices:
                state = self.states.get(idx)
                if state is not None:
                    state.reserved_bytes = max(0, state.reserved_bytes - res.bytes_per_gpu)
                    if pod_key in state.pod_keys:
                        state.pod_keys.remove(pod_key)
                    state.last_freed_at = now
            return res

    def adopt(self, pod_key: str, gpu_indices: List[int], bytes_per_gpu: int) -> None:
        """Rebuild a reservation from a pod annotation after restart, without
        schedulability checks (the pod is already running there)."""
        with self._lock:
            if pod_key in self.reservations:
                return
            for idx in gpu_indices:
                state = self.states.get(idx)
                if state is not None:
                    state.reserved_bytes += bytes_per_gpu
                    state.pod_keys.append(pod_key)
            self.reservations[pod_key] = Reservation(pod_key, list(gpu_indices), bytes_per_gpu)

    def get_reservation(self, pod_key: str) -> Optional[Reservation]:
        with self._lock:
            return self.reservations.get(pod_key)

    def schedulable_states(self) -> List[GpuState]:
        with self._lock:
            return [s for s in self.states.values() if s.schedulable]

    def total_gpus(self) -> int:
        with self._lock:
            return len(self.states)

    def schedulable_count(self) -> int:
        with self._lock:
            return sum(1 for s in self.states.values() if s.schedulable)

    def any_schedulable(self) -> bool:
        return self.schedulable_count() > 0

    def snapshot(self) -> List[GpuState]:
        with self._lock:
            return list(self.states.values())

    def set_cordoned(self, idx: int, cordoned: bool) -> bool:
        """Operator cordon/uncordon under the ledger lock (admin endpoints
        must not poke GpuState fields directly — unlocked writes raced the
        placement path). Returns False for an unknown GPU index."""
        with self._lock:
            state = self.states.get(idx)
            if state is None:
                return False
            state.cordoned = cordoned
            return True
