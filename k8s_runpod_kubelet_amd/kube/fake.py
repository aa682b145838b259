"""FakeKube — in-memory Kubernetes apiserver.

The hermetic stand-in the reference only has for the k8s side via client-go's
fake clientset (reference annotations_test.go:38). This one also implements
watches (per-watcher queues) and the pods/status strategic-merge subresource,
so the informer → controller → provider pipeline runs end to end in tests and
in bench.py without a cluster.
"""

from __future__ import annotations

import copy
import queue
import threading
import time
import uuid
from typing import Any, Dict, Iterator, List, Optional

from .client import ConflictError, GoneError, K8sClient, NotFoundError, WatchEvent
from .objects import (
    deletion_timestamp,
    meta,
    name_of,
    namespace_of,
    now_rfc3339,
)
from .patch import json_merge, strategic_merge


def _match_field_selector(pod: Dict[str, Any], selector: str) -> bool:
    if not selector:
        return True
    for clause in selector.split(","):
        if "!=" in clause:
            key, value = clause.split("!=", 1)
            negate = True
        else:
            key, value = clause.split("=", 1)
            negate = False
        actual = pod
        for part in key.strip().split("."):
            actual = actual.get(part, {}) if isinstance(actual, dict) else {}
        actual = actual if isinstance(actual, str) else ""
        if (actual == value.strip()) == negate:
            return False
    return True


class _Store:
    """One resource kind's objects + watchers + bounded event history.

    The history makes watches resumable from a resourceVersion (like the
    apiserver's watch cache): a watch with ``resourceVersion=N`` replays
    retained events with rv>N, and an rv older than the retained window is
    a 410 Gone — so informer RV-continuity/410 handling is actually
    exercised by the fake, not just code-reviewed."""

    HISTORY_LIMIT = 2048

    def __init__(self):
        self.objects: Dict[str, Dict[str, Any]] = {}  # "ns/name" or "name"
        self.watchers: List[queue.Queue] = []
        self.history: List[tuple] = []  # (rv:int, ev_type, obj)
        # highest rv ever dropped from history: a watch resuming from
        # rv < trimmed_max_rv has missed unreplayable events -> 410 Gone
        self.trimmed_max_rv = 0

    def emit(self, ev_type: str, obj: Dict[str, Any]) -> None:
        try:
            rv = int(meta(obj).get("resourceVersion", 0))
        except (TypeError, ValueError):
            rv = 0
        self.history.append((rv, ev_type, copy.deepcopy(obj)))
        if len(self.history) > self.HISTORY_LIMIT:
            drop = len(self.history) - self.HISTORY_LIMIT
            self.trimmed_max_rv = max(self.trimmed_max_rv,
                                      self.history[drop - 1][0])
            del self.history[:drop]
        for q in list(self.watchers):
            q.put((ev_type, copy.deepcopy(obj)))

    def compact(self, below_rv: int) -> None:
        """Drop history older than below_rv (etcd-compaction analogue —
        watches resuming from an older rv get 410 Gone)."""
        dropped = [h[0] for h in self.history if h[0] < below_rv]
        if dropped:
            self.trimmed_max_rv = max(self.trimmed_max_rv, max(dropped))
        self.history = [h for h in self.history if h[0] >= below_rv]


class FakeKube(K8sClient):
    def __init__(self):
        self._lock = threading.RLock()
        self._rv = 0
        self.pods = _Store()
        self.nodes = _Store()
        self.leases = _Store()
        self.secrets = _Store()
        self.configmaps = _Store()
        self.jobs = _Store()
        self.events = _Store()
        self._leases_supported = True

    # ---- helpers ----

    def _next_rv(self) -> str:
        self._rv += 1
        return str(self._rv)

    def _prepare(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        m = meta(obj)
        m.setdefault("uid", str(uuid.uuid4()))
        m.setdefault("creationTimestamp", now_rfc3339())
        m["resourceVersion"] = self._next_rv()
        return obj

    # ---- pods ----

    def list_pods(self, namespace=None, field_selector="", label_selector="") -> List[Dict[str, Any]]:
        with self._lock:
            out = []
            for pod in self.pods.objects.values():
                if namespace and namespace_of(pod) != namespace:
                    continue
                if not _match_field_selector(pod, field_selector):
                    continue
                if label_selector and not self._match_labels(pod, label_selector):
                    continue
                out.append(copy.deepcopy(pod))
            return out

    @staticmethod
    def _match_labels(obj: Dict[str, Any], selector: str) -> bool:
        lbls = meta(obj).get("labels", {})
        for clause in selector.split(","):
            key, _, value = clause.partition("=")
            if lbls.get(key.strip()) != value.strip():
                return False
        return True

    def get_pod(self, namespace: str, name: str) -> Dict[str, Any]:
        with self._lock:
            pod = self.pods.objects.get(f"{namespace}/{name}")
            if pod is None:
                raise NotFoundError(f"pod {namespace}/{name}")
            return copy.deepcopy(pod)

    def create_pod(self, namespace: str, pod: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            pod = copy.deepcopy(pod)
            meta(pod)["namespace"] = namespace
            key = f"{namespace}/{name_of(pod)}"
            if key in self.pods.objects:
                raise ConflictError(f"pod {key} exists")
            pod.setdefault("kind", "Pod")
            pod.setdefault("apiVersion", "v1")
            # apiserver-style admission defaulting (a real apiserver fills
            # these before the kubelet ever sees the pod — k8s defaults:
            # restartPolicy Always, terminationGracePeriodSeconds 30):
            spec = pod.setdefault("spec", {})
            spec.setdefault("restartPolicy", "Always")
            spec.setdefault("terminationGracePeriodSeconds", 30)
            self._prepare(pod)
            self.pods.objects[key] = pod
            self.pods.emit("ADDED", pod)
            return copy.deepcopy(pod)

    def update_pod(self, namespace: str, pod: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            key = f"{namespace}/{name_of(pod)}"
            current = self.pods.objects.get(key)
            if current is None:
                raise NotFoundError(f"pod {key}")
            sent_rv = meta(pod).get("resourceVersion")
            if sent_rv and sent_rv != meta(current)["resourceVersion"]:
                raise ConflictError(f"pod {key} resourceVersion mismatch")
            pod = copy.deepcopy(pod)
            meta(pod)["uid"] = meta(current)["uid"]
            meta(pod)["creationTimestamp"] = meta(current)["creationTimestamp"]
            if deletion_timestamp(current):
                meta(pod)["deletionTimestamp"] = deletion_timestamp(current)
            meta(pod)["resourceVersion"] = self._next_rv()
            self.pods.objects[key] = pod
            self.pods.emit("MODIFIED", pod)
            return copy.deepcopy(pod)

    def patch_pod(self, namespace: str, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            key = f"{namespace}/{name}"
            current = self.pods.objects.get(key)
            if current is None:
                raise NotFoundError(f"pod {key}")
            merged = strategic_merge(current, patch)
            meta(merged)["resourceVersion"] = self._next_rv()
            self.pods.objects[key] = merged
            self.pods.emit("MODIFIED", merged)
            return copy.deepcopy(merged)

    def patch_pod_status(self, namespace: str, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        status_patch = patch.get("status", patch)
        return self.patch_pod(namespace, name, {"status": status_patch})

    def delete_pod(self, namespace: str, name: str, grace_period_s: Optional[int] = None) -> None:
        with self._lock:
            key = f"{namespace}/{name}"
            pod = self.pods.objects.get(key)
            if pod is None:
                raise NotFoundError(f"pod {key}")
            if grace_period_s == 0 or deletion_timestamp(pod):
                # Grace 0 (or second delete of a terminating pod): remove.
                del self.pods.objects[key]
                self.pods.emit("DELETED", pod)
            else:
                meta(pod)["deletionTimestamp"] = now_rfc3339()
                meta(pod)["deletionGracePeriodSeconds"] = (
                    30 if grace_period_s is None else grace_period_s
                )
                meta(pod)["resourceVersion"] = self._next_rv()
                self.pods.emit("MODIFIED", pod)

    def list_pods_with_rv(self, namespace=None, field_selector="",
                          label_selector=""):
        with self._lock:
            return (self.list_pods(namespace=namespace,
                                   field_selector=field_selector,
                                   label_selector=label_selector),
                    str(self._rv))

    def compact_watch_history(self, below_rv: Optional[int] = None) -> None:
        """Test hook: emulate etcd compaction — watches resuming from an rv
        older than the retained window will get 410 Gone."""
        with self._lock:
            self.pods.compact(self._rv + 1 if below_rv is None else below_rv)

    def watch_pods(self, namespace=None, field_selector="", resource_version="",
                   timeout_s: float = 60.0) -> Iterator[WatchEvent]:
        q: queue.Queue = queue.Queue()
        replay: List[WatchEvent] = []
        with self._lock:
            if resource_version:
                try:
                    from_rv = int(resource_version)
                except ValueError:
                    raise GoneError(f"bad resourceVersion {resource_version!r}")
                if from_rv < self.pods.trimmed_max_rv:
                    # pod events after from_rv were dropped from the
                    # retained window: unreplayable -> 410 Gone
                    raise GoneError(
                        f"resourceVersion {from_rv} compacted "
                        f"(trimmed through {self.pods.trimmed_max_rv})")
                replay = [(t, copy.deepcopy(o))
                          for rv, t, o in self.pods.history if rv > from_rv]
            self.pods.watchers.append(q)
        deadline = time.time() + timeout_s
        try:
            for ev_type, obj in replay:
                if namespace and namespace_of(obj) != namespace:
                    continue
                if not _match_field_selector(obj, field_selector):
                    continue
                yield ev_type, obj
            while True:
                remaining = deadline - time.time()
                if remaining <= 0:
                    return
                try:
                    ev_type, obj = q.get(timeout=min(remaining, 0.5))
                except queue.Empty:
                    continue
                if namespace and namespace_of(obj) != namespace:
                    continue
                if not _match_field_selector(obj, field_selector):
                    continue
                yield ev_type, obj
        finally:
            with self._lock:
                if q in self.pods.watchers:
                    self.pods.watchers.remove(q)

    # ---- nodes ----

    def get_node(self, name: str) -> Dict[str, Any]:
        with self._lock:
            node = self.nodes.objects.get(name)
            if node is None:
                raise NotFoundError(f"node {name}")
            return copy.deepcopy(node)

    def create_node(self, node: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            key = name_of(node)
            if key in self.nodes.objects:
                raise ConflictError(f"node {key} exists")
            node = copy.deepcopy(node)
            node.setdefault("kind", "Node")
            node.setdefault("apiVersion", "v1")
            self._prepare(node)
            self.nodes.objects[key] = node
            self.nodes.emit("ADDED", node)
            return copy.deepcopy(node)

    def update_node(self, node: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            key = name_of(node)
            if key not in self.nodes.objects:
                raise NotFoundError(f"node {key}")
            node = copy.deepcopy(node)
            meta(node)["resourceVersion"] = self._next_rv()
            self.nodes.objects[key] = node
            self.nodes.emit("MODIFIED", node)
            return copy.deepcopy(node)

    def patch_node_status(self, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            current = self.nodes.objects.get(name)
            if current is None:
                raise NotFoundError(f"node {name}")
            merged = strategic_merge(current, patch)
            meta(merged)["resourceVersion"] = self._next_rv()
            self.nodes.objects[name] = merged
            self.nodes.emit("MODIFIED", merged)
            return copy.deepcopy(merged)

    def delete_node(self, name: str) -> None:
        with self._lock:
            node = self.nodes.objects.pop(name, None)
            if node is None:
                raise NotFoundError(f"node {name}")
            self.nodes.emit("DELETED", node)

    # ---- leases ----

    def set_leases_supported(self, value: bool) -> None:
        self._leases_supported = value

    def leases_supported(self) -> bool:
        return self._leases_supported

    def _lease_key(self, namespace: str, name: str) -> str:
        return f"{namespace}/{name}"

    def get_lease(self, namespace: str, name: str) -> Dict[str, Any]:
        with self._lock:
            lease = self.leases.objects.get(self._lease_key(namespace, name))
            if lease is None:
                raise NotFoundError(f"lease {namespace}/{name}")
            return copy.deepcopy(lease)

    def create_lease(self, namespace: str, lease: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            if not self._leases_supported:
                raise NotFoundError("coordination API not supported")
            lease = copy.deepcopy(lease)
            meta(lease)["namespace"] = namespace
            self._prepare(lease)
            self.leases.objects[self._lease_key(namespace, name_of(lease))] = lease
            return copy.deepcopy(lease)

    def update_lease(self, namespace: str, lease: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            key = self._lease_key(namespace, name_of(lease))
            if key not in self.leases.objects:
                raise NotFoundError(f"lease {key}")
            lease = copy.deepcopy(lease)
            meta(lease)["resourceVersion"] = self._next_rv()
            self.leases.objects[key] = lease
            return copy.deepcopy(lease)

    # ---- secrets / configmaps / jobs ----

    def _get_from(self, store: _Store, namespace: str, name: str, kind: str) -> Dict[str, Any]:
        with self._lock:
            obj = store.objects.get(f"{namespace}/{name}")
            if obj is None:
                raise NotFoundError(f"{kind} {namespace}/{name}")
            return copy.deepcopy(obj)

    def _put_into(self, store: _Store, namespace: str, obj: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            obj = copy.deepcopy(obj)
            meta(obj)["namespace"] = namespace
            self._prepare(obj)
            store.objects[f"{namespace}/{name_of(obj)}"] = obj
            return copy.deepcopy(obj)

    def get_secret(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._get_from(self.secrets, namespace, name, "secret")

    def put_secret(self, namespace: str, secret: Dict[str, Any]) -> Dict[str, Any]:
        return self._put_into(self.secrets, namespace, secret)

    def get_configmap(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._get_from(self.configmaps, namespace, name, "configmap")

    def put_configmap(self, namespace: str, cm: Dict[str, Any]) -> Dict[str, Any]:
        return self._put_into(self.configmaps, namespace, cm)

    def get_job(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._get_from(self.jobs, namespace, name, "job")

    def put_job(self, namespace: str, job: Dict[str, Any]) -> Dict[str, Any]:
        return self._put_into(self.jobs, namespace, job)

    # ---- events / auth ----

    def update_event(self, namespace: str, event: Dict[str, Any]) -> Dict[str, Any]:
        return self.create_event(namespace, event)  # name-keyed overwrite

    def create_event(self, namespace: str, event: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            event = copy.deepcopy(event)
            meta(event)["namespace"] = namespace
            self._prepare(event)
            self.events.objects[f"{namespace}/{name_of(event)}"] = event
            return copy.deepcopy(event)

    def self_subject_review(self) -> Dict[str, Any]:
        return {"status": {"userInfo": {"username": "fake-user", "groups": ["system:fake"]}}}
