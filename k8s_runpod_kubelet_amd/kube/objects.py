"""Helpers over dict-shaped Kubernetes objects.

Objects are plain dicts in the wire (JSON) shape — there is no client-go/
openapi model here; accessors below keep call sites readable and are the
single place that encodes K8s conventions (namespaced keys, RFC3339 times,
conditions-by-type)."""

from __future__ import annotations

import datetime
from typing import Any, Dict, List, Optional


def meta(obj: Dict[str, Any]) -> Dict[str, Any]:
    return obj.setdefault("metadata", {})


def name_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("name", "")


def namespace_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("namespace", "default")


def full_key(obj: Dict[str, Any]) -> str:
    """`ns/name` — the reference's deletedPods key convention (kubelet.go:628)."""
    return f"{namespace_of(obj)}/{name_of(obj)}"


def pod_key(obj: Dict[str, Any]) -> str:
    """`ns-name` — the reference's pods/podStatus map key (kubelet.go:391)."""
    return f"{namespace_of(obj)}-{name_of(obj)}"


def annotations(obj: Dict[str, Any]) -> Dict[str, str]:
    return meta(obj).setdefault("annotations", {})


def labels(obj: Dict[str, Any]) -> Dict[str, str]:
    return meta(obj).setdefault("labels", {})


def uid_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("uid", "")


def deletion_timestamp(obj: Dict[str, Any]) -> Optional[str]:
    return meta(obj).get("deletionTimestamp")


def node_name_of(pod: Dict[str, Any]) -> str:
    return pod.get("spec", {}).get("nodeName", "")


def containers_of(pod: Dict[str, Any]) -> List[Dict[str, Any]]:
    return pod.get("spec", {}).get("containers", [])


def phase_of(pod: Dict[str, Any]) -> str:
    return pod.get("status", {}).get("phase", "")


def now_rfc3339() -> str:
    return (
        datetime.datetime.now(datetime.timezone.utc)
        .replace(microsecond=0)
        .isoformat()
        .replace("+00:00", "Z")
    )


def ts_rfc3339(ts: float) -> str:
    return (
        datetime.datetime.fromtimestamp(ts, datetime.timezone.utc)
        .replace(microsecond=0)
        .isoformat()
        .replace("+00:00", "Z")
    )


def parse_rfc3339(text: str) -> float:
    return datetime.datetime.fromisoformat(text.replace("Z", "+00:00")).timestamp()


def get_condition(obj: Dict[str, Any], ctype: str) -> Optional[Dict[str, Any]]:
    for c in obj.get("status", {}).get("conditions", []):
        if c.get("type") == ctype:
            return c
    return None


def set_condition(status: Dict[str, Any], ctype: str, cstatus: str,
                  reason: str = "", message: str = "") -> None:
    conds = status.setdefault("conditions", [])
    for c in conds:
        if c.get("type") == ctype:
            if c.get("status") != cstatus:
                c["lastTransitionTime"] = now_rfc3339()
            c["status"] = cstatus
            if reason:
                c["reason"] = reason
            if message:
                c["message"] = message
            return
    cond = {"type": ctype, "status": cstatus, "lastTransitionTime": now_rfc3339()}
    if reason:
        cond["reason"] = reason
    if message:
        cond["message"] = message
    conds.append(cond)


def owner_references(obj: Dict[str, Any]) -> List[Dict[str, Any]]:
    return meta(obj).get("ownerReferences", [])


def resource_parse_cpu(value: str) -> float:
    """K8s CPU quantity → cores (e.g. "100m" → 0.1)."""
    value = str(value)
    if value.endswith("m"):
        return float(value[:-1]) / 1000.0
    return float(value)


_SUFFIX = {
    "Ki": 1024, "Mi": 1024**2, "Gi": 1024**3, "Ti": 1024**4, "Pi": 1024**5,
    "k": 1000, "M": 1000**2, "G": 1000**3, "T": 1000**4, "P": 1000**5,
}


def resource_parse_bytes(value) -> int:
    """K8s memory quantity → bytes (e.g. "256Gi")."""
    text = str(value).strip()
    for suffix, mult in _SUFFIX.items():
        if text.endswith(suffix):
            return int(float(text[: -len(suffix)]) * mult)
    return int(float(text))
