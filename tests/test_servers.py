"""Health server + kubelet API server tests (reference health.go:11-74 and
main.go:217-248 surfaces)."""

import json
import urllib.request

import pytest

from k8s_runpod_kubelet_amd.config import Config
from k8s_runpod_kubelet_amd.kube.apiserver import KubeletApiServer
from k8s_runpod_kubelet_amd.provider.provider import Provider
from k8s_runpod_kubelet_amd.runtime.fake import FakeRuntime
from k8s_runpod_kubelet_amd.server.health import HealthServer
from tests.conftest import make_pod


def get(url, expect_error=False):
    try:
        with urllib.request.urlopen(url, timeout=5) as resp:
            return resp.status, resp.read().decode()
    except urllib.error.HTTPError as exc:
        return exc.code, exc.read().decode()


@pytest.fixture
def provider(fake_kube):
    cfg = Config(notify_interval_s=0)
    rt = FakeRuntime(gpu_count=8)
    prov = Provider(fake_kube, cfg, rt)
    yield prov, rt, fake_kube
    prov.stop()


def test_health_endpoints(provider):
    prov, rt, _ = provider
    hs = HealthServer("127.0.0.1:0", prov.ping)
    hs.start()
    base = f"http://127.0.0.1:{hs.port}"
    try:
        assert get(f"{base}/healthz") == (200, "ok")
        code, _ = get(f"{base}/readyz")
        assert code == 200
        rt.set_healthy(False)
        code, body = get(f"{base}/readyz")
        assert code == 503 and "not ready" in body
        hs.set_alive(False)
        assert get(f"{base}/healthz")[0] == 503
        hs.set_alive(True)
        code, body = get(f"{base}/metrics")
        assert code == 200 and "amdvk_pod_ready_seconds" in body
        code, body = get(f"{base}/debug/threads")
        assert code == 200 and "MainThread" in body
        assert get(f"{base}/nope")[0] == 404
    finally:
        hs.stop()


def test_kubelet_api(provider, process_runtime):
    prov, _, kube = provider
    # swap in the process runtime so logs/exec are real
    prov.runtime = process_runtime
    pod = make_pod("api1", command=["podworker"], args=["--hold"])
    kube.create_pod("default", pod)
    prov.create_pod(kube.get_pod("default", "api1"))

    srv = KubeletApiServer(prov, "127.0.0.1", 0)
    srv.start()
    base = f"http://127.0.0.1:{srv.port}"
    try:
        import time
        deadline = time.time() + 5
        while time.time() < deadline:
            code, body = get(f"{base}/containerLogs/default/api1/main")
            if "podworker: ready" in body:
                break
            time.sleep(0.05)
        assert "podworker: ready" in body

        code, body = get(f"{base}/pods")
        assert code == 200
        pods = json.loads(body)
        assert pods["kind"] == "PodList"
        assert any(p["metadata"]["name"] == "api1" for p in pods["items"])

        # one-shot exec in the pod's env (reference: "not supported")
        req = urllib.request.Request(
            f"{base}/exec/default/api1/main?command=/bin/sh&command=-c&command=echo%20hi",
            method="POST")
        with urllib.request.urlopen(req, timeout=10) as resp:
            out = json.loads(resp.read())
        assert out["exitCode"] == 0
        assert "hi" in out["output"]

        assert get(f"{base}/healthz") == (200, "ok")
        code, body = get(f"{base}/stats/summary")
        assert code == 200
        summary = json.loads(body)
        assert summary["node"]["cpu"]["numCores"] >= 1
        assert summary["node"]["memory"]["usageBytes"] > 0
        api1 = [p for p in summary["pods"]
                if p["podRef"]["name"] == "api1"][0]
        # /proc-based fallback stats for the live pod process
        assert api1["memoryUsageBytes"] > 0
        assert api1["containers"][0]["name"] == "main"
    finally:
        srv.stop()
        pod_obj = prov.get_pod("default", "api1")
        if pod_obj:
            prov.delete_pod(pod_obj)


def test_cordon_endpoint(synthetic_ledger):
    """POST /cordon/<i> removes a GPU from scheduling (kubectl-cordon at
    GPU granularity); /uncordon restores it."""
    from k8s_runpod_kubelet_amd.server.health import HealthServer

    hs = HealthServer("127.0.0.1:0", None, ledger=synthetic_ledger)
    hs.start()
    base = f"http://127.0.0.1:{hs.port}"
    try:
        assert synthetic_ledger.schedulable_count() == 8
        req = urllib.request.Request(f"{base}/cordon/3", method="POST")
        assert urllib.request.urlopen(req, timeout=5).status == 200
        assert synthetic_ledger.schedulable_count() == 7
        assert not synthetic_ledger.states[3].schedulable
        req = urllib.request.Request(f"{base}/uncordon/3", method="POST")
        assert urllib.request.urlopen(req, timeout=5).status == 200
        assert synthetic_ledger.schedulable_count() == 8
        # bad index -> 404
        req = urllib.request.Request(f"{base}/cordon/99", method="POST")
        try:
            urllib.request.urlopen(req, timeout=5)
            assert False, "expected 404"
        except urllib.error.HTTPError as e:
            assert e.code == 404
    finally:
        hs.stop()


def _nonloopback_ip():
    """A local non-loopback IPv4, or None (UDP connect sends no packets)."""
    import socket

    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("10.255.255.255", 1))
        ip = s.getsockname()[0]
        return None if ip.startswith("127.") else ip
    except OSError:
        return None
    finally:
        s.close()


def test_cordon_admin_requires_loopback_or_token(synthetic_ledger):
    """Admin endpoints mutate scheduling state: non-loopback peers are
    rejected without the bearer token (round-1 advisory: anyone on the
    network could cordon every GPU)."""
    import urllib.error
    import urllib.request

    from k8s_runpod_kubelet_amd.server.health import HealthServer

    ip = _nonloopback_ip()
    if ip is None:
        import pytest

        pytest.skip("no non-loopback interface")
    hs = HealthServer("0.0.0.0:0", None, ledger=synthetic_ledger,
                      admin_token="sekrit")
    hs.start()
    try:
        base = f"http://{ip}:{hs.port}"
        req = urllib.request.Request(f"{base}/cordon/3", method="POST")
        try:
            urllib.request.urlopen(req, timeout=5)
            raise AssertionError("unauthenticated non-loopback cordon allowed")
        except urllib.error.HTTPError as exc:
            assert exc.code == 403
        assert not synthetic_ledger.states[3].cordoned
        # wrong token still rejected
        req = urllib.request.Request(f"{base}/cordon/3", method="POST",
                                     headers={"Authorization": "Bearer nope"})
        try:
            urllib.request.urlopen(req, timeout=5)
            raise AssertionError("wrong token accepted")
        except urllib.error.HTTPError as exc:
            assert exc.code == 403
        # correct token accepted
        req = urllib.request.Request(f"{base}/cordon/3", method="POST",
                                     headers={"Authorization": "Bearer sekrit"})
        assert urllib.request.urlopen(req, timeout=5).status == 200
        assert synthetic_ledger.states[3].cordoned
        # loopback works without a token (readiness for kubectl-exec'd ops)
        req = urllib.request.Request(
            f"http://127.0.0.1:{hs.port}/uncordon/3", method="POST")
        assert urllib.request.urlopen(req, timeout=5).status == 200
        assert not synthetic_ledger.states[3].cordoned
    finally:
        hs.stop()


def test_container_logs_follow_stream(provider, process_runtime):
    """`kubectl logs -f` analogue: follow=true streams appended log lines
    as they are written and terminates once the container exits."""
    import http.client
    import time

    prov, _, kube = provider
    prov.runtime = process_runtime
    # the provider subscribed to the fixture's FakeRuntime at construction;
    # re-wire events so exit status propagates from the swapped-in runtime
    process_runtime.subscribe(prov._on_runtime_event)
    pod = make_pod(
        "follower",
        command=["/bin/sh"],
        args=["-c", "echo line1; sleep 0.4; echo line2"],
    )
    kube.create_pod("default", pod)
    prov.create_pod(kube.get_pod("default", "follower"))

    srv = KubeletApiServer(prov, "127.0.0.1", 0)
    srv.start()
    try:
        conn = http.client.HTTPConnection("127.0.0.1", srv.port, timeout=20)
        t0 = time.time()
        conn.request(
            "GET", "/containerLogs/default/follower/main?follow=true")
        resp = conn.getresponse()
        assert resp.status == 200
        body = resp.read().decode()  # blocks until the stream ends
        took = time.time() - t0
        assert "line1" in body and "line2" in body
        # the stream stayed open across the sleep (really followed)
        assert took >= 0.3
        conn.close()
    finally:
        srv.stop()
        pod_obj = prov.get_pod("default", "follower")
        if pod_obj:
            prov.delete_pod(pod_obj)


def test_exec_on_pending_pod_reports_state(provider):
    """run_in_container on a pod with no instance yet must explain the pod
    state, not fail with 'instance  not found' (round-1 weak #7)."""
    prov, rt, kube = provider
    rt.deploy_error = "node full"  # deploys fail -> pod stays pending
    pod = make_pod("pender", command=["podworker"], args=["--hold"])
    kube.create_pod("default", pod)
    prov.create_pod(kube.get_pod("default", "pender"))
    code, out = prov.run_in_container("default", "pender", ["/bin/true"])
    assert code == 126
    assert "no running instance" in out
    assert "STARTING" in out


def test_exec_targets_named_container(provider, process_runtime):
    """kubectl exec -c <name>: the exec runs with the targeted container's
    credentials/confinement (multi-container pod)."""
    import urllib.request

    prov, _, kube = provider
    prov.runtime = process_runtime
    pod = make_pod("multi", containers=[
        {"name": "a", "image": "amdvk/test:latest",
         "command": ["/bin/sh"], "args": ["-c", "sleep 30"]},
        {"name": "b", "image": "amdvk/test:latest",
         "command": ["/bin/sh"], "args": ["-c", "sleep 30"],
         "securityContext": {"runAsUser": 65534, "runAsGroup": 65534}},
    ])
    kube.create_pod("default", pod)
    prov.create_pod(kube.get_pod("default", "multi"))
    srv = KubeletApiServer(prov, "127.0.0.1", 0)
    srv.start()
    try:
        base = f"http://127.0.0.1:{srv.port}"
        # container b runs as nobody; exec -c b must too
        req = urllib.request.Request(
            f"{base}/exec/default/multi/b?command=/usr/bin/id&command=-u",
            method="POST")
        with urllib.request.urlopen(req, timeout=15) as resp:
            out = json.loads(resp.read())
        assert out["exitCode"] == 0, out
        assert "65534" in out["output"], out
        # default (container a) runs as the kubelet user (root here)
        req = urllib.request.Request(
            f"{base}/exec/default/multi/a?command=/usr/bin/id&command=-u",
            method="POST")
        with urllib.request.urlopen(req, timeout=15) as resp:
            out = json.loads(resp.read())
        assert out["exitCode"] == 0, out
        assert out["output"].strip().splitlines()[-1] == "0", out
        # unknown container -> clear error
        req = urllib.request.Request(
            f"{base}/exec/default/multi/ghost?command=/bin/true",
            method="POST")
        import urllib.error

        try:
            urllib.request.urlopen(req, timeout=15)
            raise AssertionError("exec into unknown container succeeded")
        except urllib.error.HTTPError as exc:
            body = json.loads(exc.read())
            assert "not found" in body["output"]
    finally:
        srv.stop()
        pod_obj = prov.get_pod("default", "multi")
        if pod_obj:
            prov.delete_pod(pod_obj)


def test_kubelet_api_requires_auth_from_non_loopback():
    """:10250 parity: loopback is allowed, non-loopback peers need the
    bearer token (the real kubelet authenticates this surface); /healthz
    stays open for probes."""
    import socket
    import urllib.error
    import urllib.request

    ip = ""
    try:
        probe = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        probe.connect(("192.0.2.1", 9))  # no packets sent (UDP)
        ip = probe.getsockname()[0]
        probe.close()
    except OSError:
        pass
    if not ip or ip.startswith("127."):
        pytest.skip("no non-loopback interface in this environment")

    class _Stub:
        def get_pods(self):
            return []

    srv = KubeletApiServer(_Stub(), "0.0.0.0", 0, token="kubelet-secret")
    srv.start()
    try:
        base = f"http://{ip}:{srv.port}"
        # non-loopback without token: 401
        with pytest.raises(urllib.error.HTTPError) as exc:
            urllib.request.urlopen(f"{base}/pods", timeout=5)
        assert exc.value.code == 401
        # with the bearer token: allowed
        req = urllib.request.Request(
            f"{base}/pods",
            headers={"Authorization": "Bearer kubelet-secret"})
        assert urllib.request.urlopen(req, timeout=5).status == 200
        # healthz open for probes
        assert urllib.request.urlopen(
            f"{base}/healthz", timeout=5).status == 200
        # loopback needs no token
        assert urllib.request.urlopen(
            f"http://127.0.0.1:{srv.port}/pods", timeout=5).status == 200
    finally:
        srv.stop()


def test_quiet_server_swallows_client_disconnects(capsys):
    """Client hang-ups (kubectl Ctrl-C mid-stream) must not dump
    socketserver tracebacks to stderr; real errors still do."""
    from k8s_runpod_kubelet_amd.utils.httpserver import (
        QuietThreadingHTTPServer,
    )

    srv = QuietThreadingHTTPServer(("127.0.0.1", 0),
                                   __import__("http.server",
                                              fromlist=["x"]
                                              ).BaseHTTPRequestHandler)
    try:
        try:
            raise ConnectionResetError(104, "peer reset")
        except ConnectionResetError:
            srv.handle_error(None, ("127.0.0.1", 9999))
        assert "Traceback" not in capsys.readouterr().err
        try:
            raise ValueError("real bug")
        except ValueError:
            srv.handle_error(None, ("127.0.0.1", 9999))
        assert "ValueError" in capsys.readouterr().err
    finally:
        srv.server_close()


def test_container_logs_limit_bytes(provider, process_runtime):
    prov, _, kube = provider
    prov.runtime = process_runtime
    pod = make_pod("lim1", command=["/bin/sh"],
                   args=["-c", "printf 'abcdefghij%.0s' $(seq 100)"])
    kube.create_pod("default", pod)
    prov.create_pod(kube.get_pod("default", "lim1"))
    srv = KubeletApiServer(prov, "127.0.0.1", 0)
    srv.start()
    try:
        import time

        base = f"http://127.0.0.1:{srv.port}"
        deadline = time.time() + 10
        full = b""
        while time.time() < deadline:
            full = urllib.request.urlopen(
                f"{base}/containerLogs/default/lim1/main",
                timeout=5).read()
            if len(full) >= 1000:
                break
            time.sleep(0.1)
        assert len(full) == 1000
        cut = urllib.request.urlopen(
            f"{base}/containerLogs/default/lim1/main?limitBytes=64",
            timeout=5).read()
        assert cut == full[:64]
    finally:
        srv.stop()


def test_runningpods_endpoint(provider, process_runtime):
    prov, _, kube = provider
    prov.runtime = process_runtime
    pod = make_pod("rp1", command=["podworker"], args=["--hold"])
    kube.create_pod("default", pod)
    prov.create_pod(kube.get_pod("default", "rp1"))
    srv = KubeletApiServer(prov, "127.0.0.1", 0)
    srv.start()
    try:
        import time

        base = f"http://127.0.0.1:{srv.port}"
        deadline = time.time() + 10
        names = []
        while time.time() < deadline:
            body = json.loads(urllib.request.urlopen(
                f"{base}/runningpods/", timeout=5).read())
            names = [i["metadata"]["name"] for i in body["items"]]
            if "rp1" in names:
                break
            time.sleep(0.1)
        assert "rp1" in names
        assert body["kind"] == "PodList"
    finally:
        srv.stop()


def test_configz_endpoint_redacts_tokens(provider):
    prov, _, _ = provider
    prov.config.admin_token = "sekrit"
    srv = KubeletApiServer(prov, "127.0.0.1", 0)
    srv.start()
    try:
        body = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{srv.port}/configz", timeout=5).read())
        cfg = body["kubeletconfig"]
        assert cfg["node_name"] == prov.config.node_name
        assert cfg["admin_token"] == "***"
        assert "sekrit" not in json.dumps(body)
    finally:
        srv.stop()
