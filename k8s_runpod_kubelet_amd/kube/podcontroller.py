"""PodController — desired-state sync between the apiserver and the provider.

Equivalent of the virtual-kubelet library's ``node.PodController`` the
reference wires at cmd/virtual_kubelet/main.go:180-190 and runs with one
worker (main.go:263). This one runs N workers (default 4) — placement is
local and cheap, so concurrency directly improves burst drain throughput
(BASELINE config 5).

Responsibilities:
- watch pods bound to the virtual node (informer), enqueue on change,
- CreatePod / UpdatePod / DeletePod on the provider,
- finalize deletion: once the provider has torn the pod down, delete the API
  object with grace 0,
- receive provider NotifyPods callbacks and patch pod status.
"""

from __future__ import annotations

import logging
import threading
from typing import Any, Dict, Optional

from .client import K8sClient, is_not_found
from .informer import PodInformer
from .objects import deletion_timestamp, full_key, phase_of
from .workqueue import WorkQueue

log = logging.getLogger("kube.podcontroller")


class PodController:
    def __init__(
        self,
        client: K8sClient,
        informer: PodInformer,
        provider,  # provider.Provider (duck-typed to avoid import cycle)
        workers: int = 4,
    ):
        self.client = client
        self.informer = informer
        self.provider = provider
        self.workers = workers
        self.queue = WorkQueue()
        self._threads: list = []
        self._ready = threading.Event()
        self._stop = threading.Event()
        self._known_deleted: set = set()
        informer.add_handler(self._on_pod_event)

    # ---- lifecycle ----

    def start(self) -> None:
        self.informer.start()
        if not self.informer.wait_for_sync(30.0):
            raise RuntimeError("pod informer failed to sync")
        for i in range(self.workers):
            t = threading.Thread(target=self._worker, name=f"pod-worker-{i}", daemon=True)
            t.start()
            self._threads.append(t)
        self.provider.notify_pods(self._notify_from_provider)
        # deleted pods re-enqueue when their instance turns terminal, so
        # the API delete completes the moment the containers are dead
        self.provider.deletion_resync = self.queue.add
        self._ready.set()
        log.info("pod controller started", extra={"workers": self.workers})

    def ready(self) -> threading.Event:
        return self._ready

    def stop(self) -> None:
        self._stop.set()
        self.queue.shutdown()
        self.informer.stop()
        for t in self._threads:
            t.join(timeout=2.0)

    # ---- event plumbing ----

    def _on_pod_event(self, ev_type: str, pod: Dict[str, Any]) -> None:
        key = full_key(pod)
        if ev_type == "DELETED":
            self._known_deleted.add(key)
        self.queue.add(key)

    def _notify_from_provider(self, pod: Dict[str, Any]) -> None:
        """Provider pushed a status change (the event-driven path): patch the
        pods/status subresource and re-enqueue for reconciliation."""
        ns, name = full_key(pod).split("/", 1)
        try:
            self.client.patch_pod_status(ns, name, {"status": pod.get("status", {})})
        except Exception as exc:
            if not is_not_found(exc):
                log.warning("status patch from notify failed", extra={"pod": f"{ns}/{name}", "err": str(exc)})
        self.queue.add(full_key(pod))

    # ---- sync ----

    def _worker(self) -> None:
        while not self._stop.is_set():
            key = self.queue.get(timeout_s=0.5)
            if key is None:
                continue
            try:
                self._sync(key)
            except Exception:
                log.exception("pod sync failed", extra={"pod": key})
                self.queue.add_rate_limited(key)
            else:
                self.queue.forget(key)
            finally:
                self.queue.done(key)

    def _sync(self, key: str) -> None:
        ns, name = key.split("/", 1)
        pod = self.informer.get(ns, name)
        if pod is None:
            # Pod gone from the apiserver: ensure provider teardown
            # (reference cleanupDeletedPods analogue for the direct case).
            cached = self.provider.get_pod(ns, name)
            if cached is not None:
                self.provider.delete_pod(cached)
            self._known_deleted.discard(key)
            return

        if deletion_timestamp(pod):
            self.provider.delete_pod(pod)  # idempotent: record + terminate
            # k8s semantics: the API object stays Terminating until the
            # containers are actually dead — a TERM-ignoring container
            # holds the object through its grace period. The instance's
            # exit event re-enqueues this key (provider.deletion_resync);
            # informer resync is the fallback.
            if not self.provider.deletion_finalized(ns, name):
                return
            try:
                self.client.delete_pod(ns, name, grace_period_s=0)
            except Exception as exc:
                if not is_not_found(exc):
                    raise
            return

        if phase_of(pod) in ("Succeeded", "Failed"):
            return

        known = self.provider.get_pod(ns, name)
        if known is None:
            self.provider.create_pod(pod)
        else:
            self.provider.update_pod(pod)
