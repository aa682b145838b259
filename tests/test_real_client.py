"""HttpK8sClient tests via httpx.MockTransport (no cluster needed) plus
kubeconfig parsing."""

import base64
import json

import httpx
import pytest

from k8s_runpod_kubelet_amd.kube.client import ApiError, ConflictError, NotFoundError
from k8s_runpod_kubelet_amd.kube.real import (
    ClusterConfig,
    HttpK8sClient,
    create_k8s_client,
    load_kubeconfig,
)


def make_client(handler):
    client = HttpK8sClient(ClusterConfig(server="https://k8s.test", token="tok"))
    client._http = httpx.Client(
        base_url="https://k8s.test",
        headers={"Authorization": "Bearer tok"},
        transport=httpx.MockTransport(handler),
    )
    return client


def test_get_pod_path_and_auth():
    seen = {}

    def handler(request):
        seen["url"] = str(request.url)
        seen["auth"] = request.headers.get("Authorization")
        return httpx.Response(200, json={"metadata": {"name": "p1"}})

    c = make_client(handler)
    pod = c.get_pod("ns1", "p1")
    assert pod["metadata"]["name"] == "p1"
    assert seen["url"].endswith("/api/v1/namespaces/ns1/pods/p1")
    assert seen["auth"] == "Bearer tok"


def test_error_mapping():
    def handler(request):
        if "missing" in str(request.url):
            return httpx.Response(404, text="nope")
        if "conflict" in str(request.url):
            return httpx.Response(409, text="rv mismatch")
        return httpx.Response(500, text="boom")

    c = make_client(handler)
    with pytest.raises(NotFoundError):
        c.get_pod("ns", "missing")
    with pytest.raises(ConflictError):
        c.get_pod("ns", "conflict")
    with pytest.raises(ApiError) as err:
        c.get_pod("ns", "other")
    assert err.value.status_code == 500


def test_patch_content_type():
    seen = {}

    def handler(request):
        seen["ct"] = request.headers.get("Content-Type")
        seen["body"] = json.loads(request.content)
        seen["url"] = str(request.url)
        return httpx.Response(200, json={"metadata": {"name": "p"}})

    c = make_client(handler)
    c.patch_pod_status("ns", "p", {"status": {"phase": "Running"}})
    assert seen["ct"] == "application/strategic-merge-patch+json"
    assert seen["url"].endswith("/pods/p/status")
    assert seen["body"] == {"status": {"phase": "Running"}}


def test_list_pods_field_selector():
    seen = {}

    def handler(request):
        seen["params"] = dict(request.url.params)
        return httpx.Response(200, json={"items": [{"metadata": {"name": "a"}}]})

    c = make_client(handler)
    pods = c.list_pods(field_selector="spec.nodeName=n1")
    assert pods[0]["metadata"]["name"] == "a"
    assert seen["params"]["fieldSelector"] == "spec.nodeName=n1"


def test_delete_pod_grace():
    seen = {}

    def handler(request):
        seen["method"] = request.method
        seen["params"] = dict(request.url.params)
        return httpx.Response(200, json={})

    c = make_client(handler)
    c.delete_pod("ns", "p", grace_period_s=0)
    assert seen["method"] == "DELETE"
    assert seen["params"]["gracePeriodSeconds"] == "0"


def test_watch_stream_parsing():
    lines = [
        json.dumps({"type": "ADDED", "object": {"metadata": {"name": "a"}}}),
        # BOOKMARK is yielded so the informer can thread its RV forward
        json.dumps({"type": "BOOKMARK",
                    "object": {"metadata": {"resourceVersion": "7"}}}),
        json.dumps({"type": "MODIFIED", "object": {"metadata": {"name": "a"}}}),
    ]

    def handler(request):
        assert request.url.params["watch"] == "true"
        assert request.url.params["allowWatchBookmarks"] == "true"
        return httpx.Response(200, text="\n".join(lines) + "\n")

    c = make_client(handler)
    events = list(c.watch_pods(field_selector="spec.nodeName=n1", timeout_s=5))
    assert [t for t, _ in events] == ["ADDED", "BOOKMARK", "MODIFIED"]


def test_watch_stream_error_410_raises_gone():
    from k8s_runpod_kubelet_amd.kube.client import GoneError

    lines = [
        json.dumps({"type": "ADDED", "object": {"metadata": {"name": "a"}}}),
        json.dumps({"type": "ERROR",
                    "object": {"kind": "Status", "code": 410,
                               "message": "too old resource version"}}),
    ]

    def handler(request):
        return httpx.Response(200, text="\n".join(lines) + "\n")

    c = make_client(handler)
    got = []
    with pytest.raises(GoneError):
        for t, o in c.watch_pods(timeout_s=5):
            got.append(t)
    assert got == ["ADDED"]  # events before the error still delivered


def test_retry_on_transport_error():
    calls = {"n": 0}

    def handler(request):
        calls["n"] += 1
        if calls["n"] < 3:
            raise httpx.ConnectError("refused")
        return httpx.Response(200, json={"metadata": {"name": "p"}})

    c = make_client(handler)
    import k8s_runpod_kubelet_amd.utils.backoff as backoff

    orig = backoff.time.sleep
    backoff.time.sleep = lambda s: None
    try:
        pod = c.get_pod("ns", "p")
    finally:
        backoff.time.sleep = orig
    assert pod["metadata"]["name"] == "p"
    assert calls["n"] == 3  # reference retry policy: 3 attempts


def test_leases_supported_probe():
    def handler(request):
        if "coordination" in str(request.url):
            return httpx.Response(404, text="no")
        return httpx.Response(200, json={})

    c = make_client(handler)
    assert c.leases_supported() is False


def test_kubeconfig_parse(tmp_path):
    ca = base64.b64encode(b"CERTDATA").decode()
    cfg = {
        "current-context": "ctx1",
        "contexts": [{"name": "ctx1", "context": {"cluster": "c1", "user": "u1"}}],
        "clusters": [{"name": "c1", "cluster": {
            "server": "https://1.2.3.4:6443",
            "certificate-authority-data": ca}}],
        "users": [{"name": "u1", "user": {"token": "secret-token"}}],
    }
    import yaml

    path = tmp_path / "kubeconfig"
    path.write_text(yaml.safe_dump(cfg))
    cc = load_kubeconfig(str(path))
    assert cc.server == "https://1.2.3.4:6443"
    assert cc.token == "secret-token"
    assert cc.ca_path and open(cc.ca_path, "rb").read() == b"CERTDATA"


def test_create_client_no_config(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    with pytest.raises(RuntimeError, match="no Kubernetes config"):
        create_k8s_client("")


def test_in_cluster_config(tmp_path, monkeypatch):
    """In-cluster detection: serviceaccount token + env vars → https server
    with Bearer token and the mounted CA (reference createK8sClient order,
    main.go:464-502)."""
    import k8s_runpod_kubelet_amd.kube.real as real

    sa = tmp_path / "serviceaccount"
    sa.mkdir()
    (sa / "token").write_text("sekrit-token\n")
    (sa / "ca.crt").write_text("CERT")
    monkeypatch.setattr(real, "SA_DIR", str(sa))
    monkeypatch.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    monkeypatch.setenv("KUBERNETES_SERVICE_PORT", "6443")
    cc = real.load_in_cluster()
    assert cc is not None
    assert cc.server == "https://10.0.0.1:6443"
    assert cc.token == "sekrit-token"
    assert cc.ca_path.endswith("ca.crt")
    # flag beats in-cluster
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST")
    assert real.load_in_cluster() is None
