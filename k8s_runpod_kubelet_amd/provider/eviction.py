"""Node-pressure eviction (kubelet subsystem; the reference has nothing —
its pods live on remote cloud instances, so local node memory is never at
risk. Local pods ARE this node's processes, so the kubelet's
memory.available eviction signal applies for real).

Semantics follow the kubelet's hard-eviction path: when the node's
available memory drops below the configured threshold, victims are
killed immediately (no grace) and their API objects are marked
``Failed/Evicted`` — they are NOT deleted, matching ``kubectl get pods``
showing Evicted pods until a controller or human clears them.

Victim ranking (k8s eviction ordering, simplified to the signals a local
node has): BestEffort pods first, then Burstable, Guaranteed last; within
a class lower ``spec.priority`` first; ties broken by higher memory
usage. System-critical priority (>= 2e9) is exempt, like the kubelet's
``--system-critical`` guard.
"""

from __future__ import annotations

import logging
from typing import Any, Callable, Dict, List, Optional, Tuple

from .status import qos_class

log = logging.getLogger("provider.eviction")

# k8s system-cluster-critical / system-node-critical priorities
CRITICAL_PRIORITY = 2_000_000_000

_QOS_RANK = {"BestEffort": 0, "Burstable": 1, "Guaranteed": 2}


def read_available_memory_bytes(path: str = "/proc/meminfo") -> int:
    """MemAvailable from /proc/meminfo — the kubelet's memory.available
    signal source (cgroup-root variant not used: pods run in this node's
    root hierarchy alongside the host)."""
    try:
        with open(path, "r", encoding="ascii") as fh:
            for line in fh:
                if line.startswith("MemAvailable:"):
                    return int(line.split()[1]) * 1024
    except (OSError, ValueError, IndexError):
        pass
    return -1


def rank_victims(
    candidates: List[Tuple[str, Dict[str, Any], int]],
) -> List[Tuple[str, Dict[str, Any], int]]:
    """Order (pod_key, pod_obj, memory_usage_bytes) triples into eviction
    order; critical-priority pods are dropped entirely."""

    def keep(item):
        _, pod, _ = item
        prio = int(pod.get("spec", {}).get("priority", 0) or 0)
        return prio < CRITICAL_PRIORITY

    def key(item):
        _, pod, usage = item
        prio = int(pod.get("spec", {}).get("priority", 0) or 0)
        return (_QOS_RANK.get(qos_class(pod), 1), prio, -usage)

    return sorted(filter(keep, candidates), key=key)


class EvictionManager:
    """Periodic memory.available check + victim kill, driven by the
    provider's ticker. ``meminfo_reader`` is injectable for tests."""

    def __init__(self, provider, threshold_bytes: int,
                 meminfo_reader: Optional[Callable[[], int]] = None):
        self.provider = provider
        self.threshold_bytes = threshold_bytes
        self.read_available = meminfo_reader or read_available_memory_bytes

    def check(self) -> List[str]:
        """One eviction pass; returns the evicted pod keys (empty when the
        node is healthy). Evicts one pod at a time per signal crossing —
        the next tick re-reads the signal, so reclaim is measured instead
        of assumed (the kubelet's minimum-reclaim loop does the same)."""
        if self.threshold_bytes <= 0:
            return []
        avail = self.read_available()
        if avail < 0 or avail >= self.threshold_bytes:
            return []
        candidates = self.provider.eviction_candidates()
        victims = rank_victims(candidates)
        if not victims:
            log.warning("memory pressure but no evictable pods",
                        extra={"available_bytes": avail})
            return []
        key, pod, usage = victims[0]
        log.warning("evicting pod under memory pressure",
                    extra={"pod": key, "available_bytes": avail,
                           "threshold_bytes": self.threshold_bytes,
                           "pod_usage_bytes": usage})
        self.provider.evict_pod(
            key,
            f"The node was low on resource: memory. "
            f"Threshold quantity: {self.threshold_bytes} bytes, "
            f"available: {avail} bytes.")
        return [key]
