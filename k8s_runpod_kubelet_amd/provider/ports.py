"""Port extraction and the port-gated readiness check.

Counterparts of the reference's extractPortsFromPod
(pkg/virtual_kubelet/runpod_client.go:1193-1246), GetRequestedPorts
(:1379-1393) and checkPortsExposed (kubelet.go:566-605)."""

from __future__ import annotations

from typing import Any, Dict, List

from .annotations import HTTP_AUTO_PORTS, PORTS


def extract_ports_from_pod(pod: Dict[str, Any]) -> List[str]:
    """All containers' TCP containerPorts as "port/proto" strings; non-TCP
    protocols are skipped; ports in the HTTP auto-detect set become "/http"
    (runpod_client.go:1205-1235)."""
    out: List[str] = []
    seen = set()
    for container in pod.get("spec", {}).get("containers", []) or []:
        for port in container.get("ports", []) or []:
            proto = (port.get("protocol") or "TCP").upper()
            if proto != "TCP":
                continue
            try:
                number = int(port.get("containerPort", 0) or 0)
            except (TypeError, ValueError):
                continue  # malformed containerPort: skip, don't crash
            if not number or number in seen:
                continue
            seen.add(number)
            suffix = "http" if number in HTTP_AUTO_PORTS else "tcp"
            out.append(f"{number}/{suffix}")
    return out


def get_requested_ports(pod: Dict[str, Any]) -> List[str]:
    """Annotation override else spec extraction (runpod_client.go:1379-1393).
    The annotation is a comma-separated list like "8080/http,22/tcp"."""
    annotation = pod.get("metadata", {}).get("annotations", {}).get(PORTS, "")
    if annotation:
        return [p.strip() for p in annotation.split(",") if p.strip()]
    return extract_ports_from_pod(pod)


def check_ports_exposed(requested: List[str], port_mappings: Dict[int, int]) -> bool:
    """TCP ports must appear in the backend's mappings; HTTP ports are assumed
    ready (the reference assumes its HTTP proxy handles them,
    kubelet.go:583-597); no requested ports ⇒ ready (kubelet.go:570)."""
    if not requested:
        return True
    for entry in requested:
        part, _, proto = entry.partition("/")
        try:
            number = int(part)
        except ValueError:
            continue
        proto = (proto or "tcp").lower()
        if proto == "http":
            continue
        if number not in port_mappings:
            return False
    return True


def tcp_ports_of_container(container: Dict[str, Any]) -> List[int]:
    out = []
    for port in container.get("ports", []) or []:
        if (port.get("protocol") or "TCP").upper() == "TCP":
            try:
                number = int(port.get("containerPort", 0) or 0)
            except (TypeError, ValueError):
                continue  # malformed containerPort: skip, don't crash
            if number:
                out.append(number)
    return out
