"""Spec-translation tests — the param-level suite the reference gates behind
real API keys (reference annotations_test.go), here fully hermetic."""

import base64

import pytest

from k8s_runpod_kubelet_amd.config import Config
from k8s_runpod_kubelet_amd.provider import annotations as ann
from k8s_runpod_kubelet_amd.provider.envvars import extract_env_vars, is_k8s_auto_injected
from k8s_runpod_kubelet_amd.provider.ports import (
    check_ports_exposed,
    extract_ports_from_pod,
    get_requested_ports,
)
from k8s_runpod_kubelet_amd.provider.translate import (
    ValidationError,
    annotation_with_fallback,
    extract_gpu_memory_gb,
    gpu_count_of,
    prepare_deploy_params,
    validate_cloud_type,
    validate_datacenter_ids,
)
from tests.conftest import make_pod


def job_with(annotations, name="job1", uid="u-1"):
    return {"metadata": {"name": name, "namespace": "default", "uid": uid,
                         "annotations": annotations}}


def owner_ref(name="job1", uid="u-1"):
    return {"kind": "Job", "name": name, "uid": uid, "apiVersion": "batch/v1"}


# --- annotation fallback (reference annotations_test.go:27-239 scenarios) ---

def test_pod_annotation_overrides_job(fake_kube):
    fake_kube.put_job("default", job_with({ann.CLOUD_TYPE: "COMMUNITY"}))
    pod = make_pod(annotations={ann.CLOUD_TYPE: "SECURE"}, owner=owner_ref())
    from k8s_runpod_kubelet_amd.provider.translate import get_owner_job

    job = get_owner_job(pod, fake_kube)
    assert job is not None
    assert annotation_with_fallback(pod, job, ann.CLOUD_TYPE) == "SECURE"


def test_job_annotation_fallback(fake_kube):
    fake_kube.put_job("default", job_with({
        ann.REGISTRY_AUTH_ID: "auth-123",
        ann.DATACENTER_IDS: "dc1,dc2",
        ann.GPU_MEMORY: "48",
        ann.PORTS: "9999/tcp",
    }))
    pod = make_pod(owner=owner_ref())
    params = prepare_deploy_params(pod, fake_kube, Config())
    assert params.registry_auth_id == "auth-123"
    assert params.datacenter_ids == ["dc1", "dc2"]
    assert params.gpu_memory_bytes == 48 * 1024**3
    assert params.requested_ports == ["9999/tcp"]


def test_owner_job_uid_mismatch_ignored(fake_kube):
    fake_kube.put_job("default", job_with({ann.TEMPLATE_ID: "tpl"}, uid="other-uid"))
    pod = make_pod(owner=owner_ref(uid="u-1"))
    from k8s_runpod_kubelet_amd.provider.translate import get_owner_job

    assert get_owner_job(pod, fake_kube) is None


def test_template_and_auth_coexist(fake_kube):
    pod = make_pod(annotations={ann.TEMPLATE_ID: "tpl-1",
                                ann.REGISTRY_AUTH_ID: "auth-1"})
    params = prepare_deploy_params(pod, fake_kube, Config())
    assert params.template_id == "tpl-1"
    assert params.registry_auth_id == "auth-1"


# --- cloud type (runpod_client.go:1114-1134) ---

def test_cloud_type():
    assert validate_cloud_type("") == "SECURE"
    assert validate_cloud_type("secure") == "SECURE"
    assert validate_cloud_type("Community") == "COMMUNITY"
    with pytest.raises(ValidationError):
        validate_cloud_type("SPOT")


# --- datacenter allow-list (runpod_client.go:1136-1178) ---

def test_datacenter_validation():
    assert validate_datacenter_ids("", []) == []
    assert validate_datacenter_ids("a,b", []) == ["a", "b"]
    assert validate_datacenter_ids("", ["x"]) == ["x"]
    assert validate_datacenter_ids("a,x", ["x", "y"]) == ["x"]
    with pytest.raises(ValidationError):
        validate_datacenter_ids("a,b", ["x"])


# --- gpu memory (runpod_client.go:1180-1191) ---

def test_gpu_memory_parse():
    assert extract_gpu_memory_gb(make_pod(), None, 16) == 16  # default
    pod = make_pod(annotations={ann.GPU_MEMORY: "64"})
    assert extract_gpu_memory_gb(pod, None, 16) == 64
    pod = make_pod(annotations={ann.GPU_MEMORY_ALT: "256GiB"})
    assert extract_gpu_memory_gb(pod, None, 16) == 256
    pod = make_pod(annotations={ann.GPU_MEMORY: "garbage"})
    assert extract_gpu_memory_gb(pod, None, 16) == 16


def test_gpu_count_from_resources():
    assert gpu_count_of(make_pod(gpus=2)) == 2
    assert gpu_count_of(make_pod()) == 0
    pod = make_pod(containers=[
        {"name": "a", "resources": {"requests": {"amd.com/gpu": "1"}}},
        {"name": "b", "resources": {"limits": {"amd.com/gpu": "3"}}},
    ])
    assert gpu_count_of(pod) == 4


def test_memory_annotation_implies_one_gpu(fake_kube):
    pod = make_pod(annotations={ann.GPU_MEMORY: "32"})
    params = prepare_deploy_params(pod, fake_kube, Config())
    assert params.gpu_count == 1


# --- ports (runpod_client.go:1193-1246, kubelet.go:566-605) ---

def test_port_extraction_http_autodetect():
    pod = make_pod(containers=[
        {"name": "a", "ports": [
            {"containerPort": 8080, "protocol": "TCP"},
            {"containerPort": 9222},
            {"containerPort": 53, "protocol": "UDP"},  # skipped
        ]},
        {"name": "b", "ports": [{"containerPort": 443}]},
    ])
    assert extract_ports_from_pod(pod) == ["8080/http", "9222/tcp", "443/http"]


def test_requested_ports_annotation_override():
    pod = make_pod(ports=[1234], annotations={ann.PORTS: "8080/http, 22/tcp"})
    assert get_requested_ports(pod) == ["8080/http", "22/tcp"]
    pod = make_pod(ports=[1234])
    assert get_requested_ports(pod) == ["1234/tcp"]


def test_check_ports_exposed():
    assert check_ports_exposed([], {})  # no ports => ready
    assert check_ports_exposed(["80/http"], {})  # http assumed ready
    assert not check_ports_exposed(["22/tcp"], {})
    assert check_ports_exposed(["22/tcp"], {22: 22})
    assert check_ports_exposed(["22/tcp", "80/http"], {22: 10022})


# --- env extraction (runpod_client.go:866-1054) ---

def b64(s):
    return base64.b64encode(s.encode()).decode()


def test_env_literal_and_filtering(fake_kube):
    pod = make_pod(containers=[{
        "name": "a",
        "env": [
            {"name": "FOO", "value": "bar"},
            {"name": "MULTI", "value": "a\nb"},
            {"name": "KUBERNETES_SERVICE_HOST", "value": "x"},  # filtered
            {"name": "MYAPP_PORT_8080_TCP_ADDR", "value": "x"},  # filtered
        ],
    }])
    env, per = extract_env_vars(pod, fake_kube)
    assert env["FOO"] == "bar"
    assert env["MULTI"] == "a\\nb"  # newline escaping
    assert "KUBERNETES_SERVICE_HOST" not in env
    assert "MYAPP_PORT_8080_TCP_ADDR" not in env


def test_env_secret_key_ref(fake_kube):
    fake_kube.put_secret("default", {"metadata": {"name": "s1"},
                                     "data": {"tok": b64("sekrit")}})
    pod = make_pod(containers=[{
        "name": "a",
        "env": [{"name": "TOKEN",
                 "valueFrom": {"secretKeyRef": {"name": "s1", "key": "tok"}}}],
    }])
    env, _ = extract_env_vars(pod, fake_kube)
    assert env["TOKEN"] == "sekrit"


def test_env_from_whole_secret(fake_kube):
    fake_kube.put_secret("default", {"metadata": {"name": "s2"},
                                     "data": {"A": b64("1"), "B": b64("2")}})
    pod = make_pod(containers=[{"name": "a",
                                "envFrom": [{"secretRef": {"name": "s2"}}]}])
    env, _ = extract_env_vars(pod, fake_kube)
    assert env["A"] == "1" and env["B"] == "2"


def test_secret_volume_flattened(fake_kube):
    fake_kube.put_secret("default", {"metadata": {"name": "vs"},
                                     "data": {"key1": b64("v1"), "key2": b64("v2")}})
    pod = make_pod()
    pod["spec"]["volumes"] = [{"name": "v", "secret": {
        "secretName": "vs", "items": [{"key": "key1", "path": "renamed.txt"}]}}]
    env, _ = extract_env_vars(pod, fake_kube)
    assert env["renamed_txt"] == "v1"
    assert "key2" not in env
    # no items => all keys
    pod["spec"]["volumes"] = [{"name": "v", "secret": {"secretName": "vs"}}]
    env, _ = extract_env_vars(pod, fake_kube)
    assert env["key1"] == "v1" and env["key2"] == "v2"


def test_env_all_containers_parity_plus(fake_kube):
    pod = make_pod(containers=[
        {"name": "a", "env": [{"name": "X", "value": "from-a"}]},
        {"name": "b", "env": [{"name": "X", "value": "from-b"},
                              {"name": "Y", "value": "y"}]},
    ])
    env, per = extract_env_vars(pod, fake_kube)
    assert env["X"] == "from-a"  # first container wins (reference: only c0)
    assert env["Y"] == "y"
    assert per[0] == {"X": "from-a"}
    assert per[1]["X"] == "from-b"


def test_missing_secret_blocks_unless_optional(fake_kube):
    """k8s envFrom semantics: a non-optional missing secret blocks the pod
    (CreateContainerConfigError analogue: ValidationError -> Pending,
    retried); optional: true degrades to empty."""
    pod = make_pod(containers=[{"name": "a",
                                "envFrom": [{"secretRef": {"name": "nope"}}]}])
    with pytest.raises(ValidationError, match="missing secret"):
        extract_env_vars(pod, fake_kube)
    pod = make_pod(containers=[{
        "name": "a",
        "envFrom": [{"secretRef": {"name": "nope", "optional": True}}]}])
    env, _ = extract_env_vars(pod, fake_kube)
    assert env == {}
    # single-key refs: same rule
    pod = make_pod(containers=[{
        "name": "a",
        "env": [{"name": "X", "valueFrom": {
            "secretKeyRef": {"name": "nope", "key": "k"}}}]}])
    with pytest.raises(ValidationError, match="missing secret"):
        extract_env_vars(pod, fake_kube)
    pod = make_pod(containers=[{
        "name": "a",
        "env": [{"name": "X", "valueFrom": {
            "secretKeyRef": {"name": "nope", "key": "k",
                             "optional": True}}}]}])
    env, _ = extract_env_vars(pod, fake_kube)
    assert "X" not in env


def test_auto_injected_patterns():
    assert is_k8s_auto_injected("KUBERNETES_PORT")
    assert is_k8s_auto_injected("FOO_SERVICE_HOST")
    assert not is_k8s_auto_injected("MY_TOKEN")


# --- full assembly (runpod_client.go:1248-1377) ---

def test_prepare_full(fake_kube):
    cfg = Config(datacenter_ids=["dc1"])
    pod = make_pod(gpus=2, ports=[8080],
                   annotations={ann.GPU_MEMORY: "128", ann.CLOUD_TYPE: "secure"})
    pod["spec"]["containers"][0]["resources"]["limits"].update(
        {"cpu": "2", "memory": "4Gi"})
    params = prepare_deploy_params(pod, fake_kube, cfg)
    assert params.pod_key == "default-p1"
    assert params.gpu_count == 2
    assert params.gpu_memory_bytes == 128 * 1024**3
    assert params.cloud_type == "SECURE"
    assert params.datacenter_ids == ["dc1"]
    assert params.requested_ports == ["8080/http"]
    assert params.cpu_limit == "200000 100000"
    assert params.memory_limit == str(4 * 1024**3)
    assert params.max_gpu_cost == 0.5


def test_prepare_no_containers_rejected(fake_kube):
    pod = make_pod()
    pod["spec"]["containers"] = []
    with pytest.raises(ValidationError):
        prepare_deploy_params(pod, fake_kube, Config())


def test_prepare_fail_fast_when_no_offers(fake_kube, synthetic_ledger):
    from k8s_runpod_kubelet_amd.provider.selector import GpuOfferCatalog

    catalog = GpuOfferCatalog(synthetic_ledger)
    pod = make_pod(gpus=1, annotations={ann.GPU_MEMORY: "10000"})  # > 288 GB
    with pytest.raises(ValidationError, match="no GPU set available"):
        prepare_deploy_params(pod, fake_kube, Config(), catalog)
    ok = make_pod(gpus=8, annotations={ann.GPU_MEMORY: "2000"})  # 250/GPU
    params = prepare_deploy_params(ok, fake_kube, Config(), catalog)
    assert params.gpu_count == 8


def test_configmap_and_fieldref_env(fake_kube, pod_factory):
    """ConfigMap-sourced env + Downward-API fieldRef (parity-plus: the
    reference supports Secret sources only, runpod_client.go:866-1054)."""
    from k8s_runpod_kubelet_amd.provider.envvars import extract_env_vars

    fake_kube.put_configmap("default", {
        "metadata": {"name": "app-config", "namespace": "default"},
        "data": {"MODE": "prod", "THREADS": "8"},
    })
    pod = pod_factory("cmenv", containers=[{
        "name": "main", "image": "x",
        "envFrom": [{"configMapRef": {"name": "app-config"}}],
        "env": [
            {"name": "ONE_KEY",
             "valueFrom": {"configMapKeyRef": {"name": "app-config",
                                               "key": "MODE"}}},
            {"name": "MY_POD",
             "valueFrom": {"fieldRef": {"fieldPath": "metadata.name"}}},
            {"name": "MY_NODE",
             "valueFrom": {"fieldRef": {"fieldPath": "spec.nodeName"}}},
        ],
    }])
    pod_env, per = extract_env_vars(pod, fake_kube)
    assert pod_env["MODE"] == "prod" and pod_env["THREADS"] == "8"
    assert pod_env["ONE_KEY"] == "prod"
    assert pod_env["MY_POD"] == "cmenv"
    assert pod_env["MY_NODE"] == "virtual-runpod"
    # missing configmap: optional degrades to empty, non-optional blocks
    pod2 = pod_factory("cm2", containers=[{
        "name": "main", "image": "x",
        "envFrom": [{"configMapRef": {"name": "nope", "optional": True}}],
    }])
    pod_env2, _ = extract_env_vars(pod2, fake_kube)
    assert "MODE" not in pod_env2
    pod3 = pod_factory("cm3", containers=[{
        "name": "main", "image": "x",
        "envFrom": [{"configMapRef": {"name": "nope"}}],
    }])
    with pytest.raises(ValidationError, match="missing configMap"):
        extract_env_vars(pod3, fake_kube)


def test_projected_volume_merges_sources(fake_kube):
    """projected volume: secret + configMap + downwardAPI sources land in
    one files map; items[] renames apply per source; serviceAccountToken
    projects a (possibly empty, offline) token file."""
    from k8s_runpod_kubelet_amd.provider.translate import extract_volumes

    fake_kube.put_secret("default", {"metadata": {"name": "ps"},
                                     "data": {"tok": b64("s3cr3t")}})
    fake_kube.put_configmap("default", {"metadata": {"name": "pc"},
                                         "data": {"conf": "cfg-val"}})
    pod = make_pod()
    pod["metadata"]["labels"] = {"app": "demo"}
    pod["spec"]["volumes"] = [{"name": "proj", "projected": {
        "defaultMode": 0o600,
        "sources": [
            {"secret": {"name": "ps",
                        "items": [{"key": "tok", "path": "token.txt"}]}},
            {"configMap": {"name": "pc"}},
            {"downwardAPI": {"items": [
                {"path": "labels", "fieldRef": {"fieldPath":
                                                "metadata.labels"}}]}},
            {"serviceAccountToken": {"path": "sa-token",
                                     "audience": "api"}},
        ]}}]
    vols = extract_volumes(pod, fake_kube)
    v = vols["proj"]
    assert v.kind == "files"
    assert v.file_mode == 0o600
    assert v.files["token.txt"] == "s3cr3t"
    assert v.files["conf"] == "cfg-val"
    assert 'app="demo"' in v.files["labels"]
    assert "sa-token" in v.files  # best-effort: empty offline


def test_container_pull_and_termination_policies_parsed(fake_kube):
    pod = make_pod(containers=[{
        "name": "a", "image": "example/x:latest",
        "imagePullPolicy": "Never",
        "terminationMessagePolicy": "FallbackToLogsOnError",
    }])
    params = prepare_deploy_params(pod, fake_kube, Config())
    c = params.containers[0]
    assert c.image_pull_policy == "Never"
    assert c.termination_message_policy == "FallbackToLogsOnError"


def test_volume_secret_optional_and_missing(fake_kube):
    from k8s_runpod_kubelet_amd.provider.translate import (ValidationError,
                                                           extract_volumes)

    pod = make_pod()
    pod["spec"]["volumes"] = [{"name": "v", "secret": {
        "secretName": "ghost"}}]
    with pytest.raises(ValidationError, match="not found"):
        extract_volumes(pod, fake_kube)
    pod["spec"]["volumes"] = [{"name": "v", "secret": {
        "secretName": "ghost", "optional": True}}]
    vols = extract_volumes(pod, fake_kube)
    assert vols["v"].kind == "files" and vols["v"].files == {}
    # configMap analogue
    pod["spec"]["volumes"] = [{"name": "v", "configMap": {"name": "nope"}}]
    with pytest.raises(ValidationError, match="not found"):
        extract_volumes(pod, fake_kube)


def test_empty_dir_medium_and_size_parsed(fake_kube):
    from k8s_runpod_kubelet_amd.provider.translate import extract_volumes

    pod = make_pod()
    pod["spec"]["volumes"] = [
        {"name": "mem", "emptyDir": {"medium": "Memory",
                                     "sizeLimit": "64Mi"}},
        {"name": "disk", "emptyDir": {}},
    ]
    vols = extract_volumes(pod, fake_kube)
    assert vols["mem"].medium == "Memory"
    assert vols["mem"].size_limit_bytes == 64 << 20
    assert vols["disk"].medium == ""


def test_dns_config_rendered(fake_kube):
    pod = make_pod()
    pod["spec"]["dnsConfig"] = {
        "nameservers": ["10.0.0.10", "10.0.0.11"],
        "searches": ["ns.svc.cluster.local", "example.com"],
        "options": [{"name": "ndots", "value": "5"}, {"name": "edns0"}],
    }
    params = prepare_deploy_params(pod, fake_kube, Config())
    assert params.resolv_conf == (
        "nameserver 10.0.0.10\nnameserver 10.0.0.11\n"
        "search ns.svc.cluster.local example.com\n"
        "options ndots:5 edns0\n")
    # no dnsConfig -> node default (empty marker)
    assert prepare_deploy_params(make_pod(), fake_kube,
                                 Config()).resolv_conf == ""
