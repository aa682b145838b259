"""NodeController — registers the virtual node and keeps it fresh.

Equivalent of virtual-kubelet's ``node.NodeController`` with the optional
coordination-v1 Lease wired the same way the reference does
(reference cmd/virtual_kubelet/main.go:193-213: lease interval 30 s if the
coordination API is discoverable, graceful fallback to status-only
otherwise)."""

from __future__ import annotations

import logging
import threading
from typing import Any, Dict

from ..utils.backoff import Ticker
from .client import K8sClient, is_not_found
from .objects import meta, now_rfc3339

log = logging.getLogger("kube.nodecontroller")

LEASE_INTERVAL_S = 30.0
LEASE_NS = "kube-node-lease"


class NodeController:
    def __init__(
        self,
        client: K8sClient,
        provider,  # NodeProvider surface: get_node_status(), ping(), notify_node_status(cb)
        status_interval_s: float = 30.0,
    ):
        self.client = client
        self.provider = provider
        self.status_interval_s = status_interval_s
        self._lease_enabled = False
        self._tickers: list = []
        self._node_name = ""

    def start(self) -> None:
        node = self.provider.get_node_status()
        self._node_name = meta(node)["name"]
        self._register(node)
        self._lease_enabled = self.client.leases_supported()
        if self._lease_enabled:
            try:
                self._renew_lease()
            except Exception:
                log.warning("lease creation failed; falling back to status-only pings")
                self._lease_enabled = False

        self.provider.notify_node_status(self._push_status)
        self._tickers.append(
            Ticker(self.status_interval_s, self._periodic_status, "node-status").start()
        )
        if self._lease_enabled:
            self._tickers.append(
                Ticker(LEASE_INTERVAL_S, self._renew_lease_safe, "node-lease").start()
            )
        log.info("node controller started", extra={"node": self._node_name, "leases": self._lease_enabled})

    def stop(self) -> None:
        for t in self._tickers:
            t.stop()

    # ---- internals ----

    def _register(self, node: Dict[str, Any]) -> None:
        try:
            existing = self.client.get_node(self._node_name)
        except Exception as exc:
            if not is_not_found(exc):
                raise
            self.client.create_node(node)
            log.info("registered virtual node", extra={"node": self._node_name})
            return
        # Adopt: keep metadata, refresh spec/status.
        existing["spec"] = node.get("spec", {})
        existing["status"] = node.get("status", {})
        merged_meta = existing.setdefault("metadata", {})
        merged_meta.setdefault("labels", {}).update(meta(node).get("labels", {}))
        self.client.update_node(existing)
        log.info("adopted existing virtual node", extra={"node": self._node_name})

    def _push_status(self, node: Dict[str, Any]) -> None:
        try:
            self.client.patch_node_status(self._node_name, {"status": node.get("status", {})})
        except Exception as exc:
            if is_not_found(exc):
                # Node deleted out from under us: re-register.
                self._register(self.provider.get_node_status())
            else:
                log.warning("node status patch failed", extra={"err": str(exc)})

    def _periodic_status(self) -> None:
        try:
            self.provider.ping()
            node = self.provider.get_node_status()
        except Exception as exc:
            node = self.provider.get_node_status()
            for cond in node.get("status", {}).get("conditions", []):
                if cond.get("type") == "Ready":
                    cond["status"] = "False"
                    cond["reason"] = "ProviderPingFailed"
                    cond["message"] = str(exc)
        self._push_status(node)

    def _renew_lease_safe(self) -> None:
        try:
            self._renew_lease()
        except Exception:
            log.exception("lease renewal failed")

    def _renew_lease(self) -> None:
        lease = {
            "apiVersion": "coordination.k8s.io/v1",
            "kind": "Lease",
            "metadata": {"name": self._node_name, "namespace": LEASE_NS},
            "spec": {
                "holderIdentity": self._node_name,
                "leaseDurationSeconds": int(LEASE_INTERVAL_S * 2),
                "renewTime": now_rfc3339(),
            },
        }
        try:
            current = self.client.get_lease(LEASE_NS, self._node_name)
            current["spec"] = lease["spec"]
            self.client.update_lease(LEASE_NS, current)
        except Exception as exc:
            if not is_not_found(exc):
                raise
            self.client.create_lease(LEASE_NS, lease)
