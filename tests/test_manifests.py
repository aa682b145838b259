"""Deployment artifacts sanity: raw manifest parses and carries the RBAC
the provider needs (reference deploy/kubelet.yaml:1-103 contract)."""

import os

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_raw_manifest_parses_and_rbac_complete():
    docs = list(yaml.safe_load_all(
        open(os.path.join(REPO, "deploy", "kubelet.yaml"))))
    kinds = {d["kind"] for d in docs if d}
    assert {"ClusterRole", "ClusterRoleBinding", "ServiceAccount",
            "Deployment"} <= kinds
    role = next(d for d in docs if d and d["kind"] == "ClusterRole")
    rules = {r: set() for r in ("pods", "nodes", "secrets", "configmaps",
                                "events", "leases", "pods/status")}
    for rule in role["rules"]:
        for res in rule.get("resources", []):
            if res in rules:
                rules[res] |= set(rule.get("verbs", []))
    # verbs the provider actually uses
    assert {"get", "list", "watch"} <= rules["pods"]
    assert {"update", "patch"} & rules["pods/status"]
    assert {"create", "update"} & rules["nodes"] or "*" in rules["nodes"]
    assert {"get"} <= rules["secrets"]
    assert {"create"} <= rules["events"]
    assert {"create", "update"} & rules["leases"] or "*" in rules["leases"]
    dep = next(d for d in docs if d and d["kind"] == "Deployment")
    ctr = dep["spec"]["template"]["spec"]["containers"][0]
    mounts = {m["mountPath"] for m in ctr["volumeMounts"]}
    assert {"/dev/kfd", "/dev/dri"} <= mounts  # local GPU backend needs devices


def test_helm_values_parse_and_keep_reference_keys():
    values = yaml.safe_load(open(os.path.join(
        REPO, "helm", "amd-virtual-kubelet", "values.yaml")))
    # reference-compatible keys (helm/runpod-kubelet/values.yaml)
    assert "reconcileInterval" in values["kubelet"]
    assert "maxGpuPrice" in values["kubelet"]
    assert "healthServerAddress" in values["kubelet"]
    assert "identifier" in values["cluster"]
    assert "apiToken" in values["conduit"]
