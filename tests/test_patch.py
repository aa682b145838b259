from k8s_runpod_kubelet_amd.kube.patch import json_merge, strategic_merge


def test_map_merge_recursive():
    orig = {"a": {"b": 1, "c": 2}, "d": 3}
    patch = {"a": {"b": 9}, "e": 4}
    assert strategic_merge(orig, patch) == {"a": {"b": 9, "c": 2}, "d": 3, "e": 4}


def test_none_deletes_key():
    assert strategic_merge({"a": 1, "b": 2}, {"a": None}) == {"b": 2}


def test_conditions_merge_by_type():
    orig = {"conditions": [{"type": "Ready", "status": "False", "reason": "x"},
                           {"type": "PodScheduled", "status": "True"}]}
    patch = {"conditions": [{"type": "Ready", "status": "True"}]}
    merged = strategic_merge(orig, patch)
    by_type = {c["type"]: c for c in merged["conditions"]}
    assert by_type["Ready"]["status"] == "True"
    assert by_type["Ready"]["reason"] == "x"  # merged, not replaced
    assert "PodScheduled" in by_type


def test_container_statuses_merge_by_name():
    orig = {"containerStatuses": [{"name": "a", "restartCount": 1}]}
    patch = {"containerStatuses": [{"name": "a", "ready": True},
                                   {"name": "b", "ready": False}]}
    merged = strategic_merge(orig, patch)
    assert merged["containerStatuses"][0] == {"name": "a", "restartCount": 1,
                                              "ready": True}
    assert len(merged["containerStatuses"]) == 2


def test_unknown_list_replaced():
    orig = {"finalizers": ["a", "b"]}
    patch = {"finalizers": ["c"]}
    assert strategic_merge(orig, patch)["finalizers"] == ["c"]


def test_json_merge():
    assert json_merge({"a": {"b": 1}}, {"a": {"c": 2}}) == {"a": {"b": 1, "c": 2}}
    assert json_merge({"a": 1}, {"a": None}) == {}
