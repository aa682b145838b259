"""Event recorder — the EventBroadcaster analogue (reference
cmd/virtual_kubelet/main.go:172-177 wires a client-go recorder; the
virtual-kubelet lib emits through it)."""

from __future__ import annotations

import logging
import time
from typing import Any, Dict

from .client import K8sClient
from .objects import name_of, namespace_of, now_rfc3339, uid_of

log = logging.getLogger("kube.events")


class EventRecorder:
    def __init__(self, client: K8sClient, component: str = "amd-virtual-kubelet"):
        self.client = client
        self.component = component

    def event(self, obj: Dict[str, Any], event_type: str, reason: str,
              message: str) -> None:
        namespace = namespace_of(obj)
        event = {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {
                "name": f"{name_of(obj)}.{int(time.time() * 1e6):x}",
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
            "firstTimestamp": now_rfc3339(),
            "lastTimestamp": now_rfc3339(),
            "count": 1,
        }
        try:
            self.client.create_event(namespace, event)
        except Exception as exc:
            log.debug("event create failed", extra={"err": str(exc)})
