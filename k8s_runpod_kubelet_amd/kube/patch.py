"""Strategic-merge-patch — enough of it for pod/status and node/status.

The reference PATCHes the pods/status subresource with a strategic merge
patch (reference kubelet.go:1822-1845 updatePodStatusInK8s). K8s SMP merges
maps recursively and merges certain lists by a per-type merge key; the table
below covers the lists this kubelet touches (conditions by ``type``,
containerStatuses by ``name``, etc.). Unknown lists are replaced, matching
SMP's default for lists without a patchStrategy.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional

MERGE_KEYS = {
    "conditions": "type",
    "containerStatuses": "name",
    "initContainerStatuses": "name",
    "ephemeralContainerStatuses": "name",
    "containers": "name",
    "volumes": "name",
    "env": "name",
    "ports": "containerPort",
    "addresses": "type",
    "tolerations": "key",
    "images": None,  # replace
}


def strategic_merge(original: Any, patch: Any, key_hint: Optional[str] = None) -> Any:
    if patch is None:
        return None
    if isinstance(patch, dict) and isinstance(original, dict):
        out = copy.deepcopy(original)
        for key, value in patch.items():
            if value is None:
                out.pop(key, None)
            elif key in out:
                out[key] = strategic_merge(out[key], value, key_hint=key)
            else:
                out[key] = copy.deepcopy(value)
        return out
    if isinstance(patch, list) and isinstance(original, list) and key_hint in MERGE_KEYS:
        merge_key = MERGE_KEYS[key_hint]
        if merge_key is None:
            return copy.deepcopy(patch)
        out: List[Any] = copy.deepcopy(original)
        index = {
            item.get(merge_key): i
            for i, item in enumerate(out)
            if isinstance(item, dict)
        }
        for item in patch:
            if isinstance(item, dict) and item.get(merge_key) in index:
                i = index[item[merge_key]]
                out[i] = strategic_merge(out[i], item)
            else:
                out.append(copy.deepcopy(item))
        return out
    return copy.deepcopy(patch)


def json_merge(original: Any, patch: Any) -> Any:
    """RFC 7386 merge patch (used for annotations-only patches)."""
    if not isinstance(patch, dict):
        return copy.deepcopy(patch)
    out = copy.deepcopy(original) if isinstance(original, dict) else {}
    for key, value in patch.items():
        if value is None:
            out.pop(key, None)
        else:
            out[key] = json_merge(out.get(key), value)
    return out
