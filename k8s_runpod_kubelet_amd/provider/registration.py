"""Optional registration + heartbeat hook.

Surface-shape parity with the reference's Conduit registration
(pkg/virtual_kubelet/kubelet.go:54-289: PUT /api/kubelet/register with a
cluster/node payload, re-sent every heartbeat interval). Deliberately NOT a
licensing gate: the reference makes registration mandatory and fails provider
construction on error (kubelet.go:369-371); here it is off unless a
``registration_endpoint`` is configured, and failures only log (SURVEY §7
non-goals)."""

from __future__ import annotations

import logging
import os
import threading
from typing import Dict, Optional

from ..utils.backoff import Ticker
from ..version import __version__

log = logging.getLogger("provider.registration")


def build_payload(node_name: str, namespace: str, gpu_count: int) -> Dict:
    """Payload fields mirror the reference RegistrationPayload
    (kubelet.go:88-127)."""
    return {
        "clusterName": os.environ.get("CLUSTER_NAME", "unknown"),
        "namespace": namespace,
        "nodeName": node_name,
        "version": __version__,
        "capabilities": ["gpu-pods", "amd.com/gpu", "event-driven-status"],
        "metadata": {"provider": "amd-mi355x", "gpuCount": gpu_count},
    }


class Registrar:
    def __init__(self, endpoint: str, node_name: str, namespace: str,
                 gpu_count: int, heartbeat_interval_s: float = 300.0):
        self.endpoint = endpoint.rstrip("/") if endpoint else ""
        self.node_name = node_name
        self.namespace = namespace
        self.gpu_count = gpu_count
        self.heartbeat_interval_s = heartbeat_interval_s
        self._ticker: Optional[Ticker] = None
        self._lock = threading.Lock()
        self.registered = False

    def register(self) -> bool:
        if not self.endpoint:
            return True  # registration disabled — always "registered"
        ok = self._send()
        if ok and self.heartbeat_interval_s > 0:
            self.start_heartbeat()
        return ok

    def _send(self) -> bool:
        import httpx

        token = os.environ.get("CONDUIT_API_TOKEN", "")
        headers = {"Content-Type": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        try:
            resp = httpx.put(
                f"{self.endpoint}/api/kubelet/register",
                json=build_payload(self.node_name, self.namespace, self.gpu_count),
                headers=headers,
                timeout=10.0,
            )
            ok = resp.status_code < 300
            with self._lock:
                self.registered = ok
            if not ok:
                log.warning("registration rejected", extra={"status": resp.status_code})
            return ok
        except Exception as exc:
            log.warning("registration failed", extra={"err": str(exc)})
            return False

    def start_heartbeat(self) -> None:
        """Heartbeat re-sends the registration every interval (reference
        sendHeartbeat/startHeartbeat, kubelet.go:165-255); 0 disables
        (kubelet.go:73)."""
        if self._ticker is None and self.heartbeat_interval_s > 0 and self.endpoint:
            self._ticker = Ticker(self.heartbeat_interval_s, self._send, "heartbeat").start()

    def stop_heartbeat(self) -> None:
        """StopHeartbeat analogue (kubelet.go:282-289)."""
        if self._ticker is not None:
            self._ticker.stop()
            self._ticker = None
