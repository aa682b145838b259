"""Event recorder — the EventBroadcaster analogue (reference
cmd/virtual_kubelet/main.go:172-177 wires a client-go recorder; the
virtual-kubelet lib emits through it)."""

from __future__ import annotations

import logging
import threading
import time
from typing import Any, Dict

from .client import K8sClient
from .objects import name_of, namespace_of, now_rfc3339, uid_of

log = logging.getLogger("kube.events")


class EventRecorder:
    """client-go-style correlation: a repeat of the same
    (object, type, reason, message) within the aggregation window bumps
    the existing Event's count/lastTimestamp instead of creating a new
    object — a CrashLoopBackOff pod otherwise floods the apiserver with
    one BackOff Event per restart attempt."""

    AGGREGATION_WINDOW_S = 600.0
    _CACHE_MAX = 4096

    def __init__(self, client: K8sClient, component: str = "amd-virtual-kubelet"):
        self.client = client
        self.component = component
        # (ns, name, uid, type, reason, message) -> [event_name, count,
        # first_ts, last_seen_monotonic]
        self._seen: Dict[tuple, list] = {}
        self._lock = threading.Lock()

    def event(self, obj: Dict[str, Any], event_type: str, reason: str,
              message: str) -> None:
        namespace = namespace_of(obj)
        key = (namespace, name_of(obj), uid_of(obj), event_type, reason,
               message)
        now_mono = time.monotonic()
        with self._lock:  # events fire from deploy/reconcile/event threads
            hit = self._seen.get(key)
            if hit is not None and \
                    now_mono - hit[3] < self.AGGREGATION_WINDOW_S:
                hit[1] += 1
                hit[3] = now_mono
                ev_name, count, first_ts = hit[0], hit[1], hit[2]
            else:
                ev_name = f"{name_of(obj)}.{int(time.time() * 1e6):x}"
                first_ts = now_rfc3339()
                count = 1
                if len(self._seen) >= self._CACHE_MAX:
                    # drop the stalest half; bounded memory under churn
                    by_age = sorted(self._seen.items(),
                                    key=lambda kv: kv[1][3])
                    for k, _ in by_age[: self._CACHE_MAX // 2]:
                        self._seen.pop(k, None)
                self._seen[key] = [ev_name, 1, first_ts, now_mono]
        event = {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {
                "name": ev_name,
                "namespace": namespace,
            },
            "involvedObject": {
                "kind": obj.get("kind", "Pod"),
                "namespace": namespace,
                "name": name_of(obj),
                "uid": uid_of(obj),
            },
            "reason": reason,
            "message": message,
            "type": event_type,
            "source": {"component": self.component},
            "firstTimestamp": first_ts,
            "lastTimestamp": now_rfc3339(),
            "count": count,
        }
        try:
            if count > 1:
                update = getattr(self.client, "update_event", None)
                if callable(update):
                    try:
                        update(namespace, event)
                        return
                    except Exception:
                        # apiserver TTL-GCs Events (~1 h): the aggregated
                        # object may be gone — restart the series under a
                        # fresh name rather than losing the signal (a
                        # fresh name also survives non-404 update
                        # failures without clobbering a live object)
                        ev_name = (f"{name_of(obj)}."
                                   f"{int(time.time() * 1e6):x}")
                        first_ts = now_rfc3339()
                        with self._lock:
                            self._seen[key] = [ev_name, 1, first_ts,
                                               now_mono]
                        event["metadata"]["name"] = ev_name
                        event["count"] = 1
                        event["firstTimestamp"] = first_ts
            self.client.create_event(namespace, event)
        except Exception as exc:
            log.debug("event create failed", extra={"err": str(exc)})
