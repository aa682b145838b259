"""Pod informer: list + watch with periodic resync.

Counterpart of the reference's SharedInformerFactory setup
(reference cmd/virtual_kubelet/main.go:147-164): pods filtered by the field
selector ``spec.nodeName==<node>``, resync driven by the configured reconcile
interval — which here actually re-enqueues everything (in the reference the
interval only sets informer resync while the real loops are hardcoded)."""

from __future__ import annotations

import logging
import threading
from typing import Any, Callable, Dict, List, Optional

from .client import K8sClient
from .objects import full_key

log = logging.getLogger("kube.informer")

Handler = Callable[[str, Dict[str, Any]], None]  # (event_type, pod)


class PodInformer:
    def __init__(
        self,
        client: K8sClient,
        node_name: str,
        resync_interval_s: float = 30.0,
        namespace: Optional[str] = None,
    ):
        self.client = client
        self.node_name = node_name
        self.namespace = namespace
        self.resync_interval_s = resync_interval_s
        self.field_selector = f"spec.nodeName={node_name}"
        self._handlers: List[Handler] = []
        self._cache: Dict[str, Dict[str, Any]] = {}
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self._synced = threading.Event()
        self._threads: List[threading.Thread] = []

    def add_handler(self, handler: Handler) -> None:
        self._handlers.append(handler)

    def start(self) -> None:
        t_watch = threading.Thread(target=self._run_watch, name="pod-informer", daemon=True)
        t_resync = threading.Thread(target=self._run_resync, name="pod-resync", daemon=True)
        self._threads = [t_watch, t_resync]
        for t in self._threads:
            t.start()

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2.0)

    def wait_for_sync(self, timeout_s: float = 10.0) -> bool:
        return self._synced.wait(timeout_s)

    # ---- cache access ----

    def get(self, namespace: str, name: str) -> Optional[Dict[str, Any]]:
        with self._lock:
            return self._cache.get(f"{namespace}/{name}")

    def list(self) -> List[Dict[str, Any]]:
        with self._lock:
            return list(self._cache.values())

    # ---- internals ----

    def _dispatch(self, ev_type: str, pod: Dict[str, Any]) -> None:
        for handler in self._handlers:
            try:
                handler(ev_type, pod)
            except Exception:
                log.exception("informer handler failed")

    def _full_list(self) -> None:
        pods = self.client.list_pods(
            namespace=self.namespace, field_selector=self.field_selector
        )
        with self._lock:
            seen = set()
            for pod in pods:
                key = full_key(pod)
                seen.add(key)
                existed = key in self._cache
                self._cache[key] = pod
                self._dispatch("MODIFIED" if existed else "ADDED", pod)
            for key in list(self._cache):
                if key not in seen:
                    gone = self._cache.pop(key)
                    self._dispatch("DELETED", gone)

    def _run_watch(self) -> None:
        while not self._stop.is_set():
            try:
                self._full_list()
                self._synced.set()
                for ev_type, pod in self.client.watch_pods(
                    namespace=self.namespace,
                    field_selector=self.field_selector,
                    timeout_s=30.0,
                ):
                    if self._stop.is_set():
                        return
                    key = full_key(pod)
                    with self._lock:
                        if ev_type == "DELETED":
                            self._cache.pop(key, None)
                        else:
                            self._cache[key] = pod
                    self._dispatch(ev_type, pod)
            except Exception:
                if self._stop.is_set():
                    return
                log.exception("pod watch failed; relisting")
                self._stop.wait(1.0)

    def _run_resync(self) -> None:
        while not self._stop.wait(self.resync_interval_s):
            try:
                self._full_list()
            except Exception:
                log.exception("resync list failed")
