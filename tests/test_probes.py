"""Probe unit tests + e2e liveness/readiness (the reference has no probe
support at all; readiness there is inferred from cloud port mappings)."""

import socket
import threading
import time

import pytest

from k8s_runpod_kubelet_amd.runtime.probes import (
    ProbeSpec, ProbeState, advance, run_probe)


def test_parse_probe_kinds():
    tcp = ProbeSpec.parse({"tcpSocket": {"port": 8080},
                           "periodSeconds": 2, "failureThreshold": 5})
    assert tcp.kind == "tcp" and tcp.port == 8080
    assert tcp.period_s == 2 and tcp.failure_threshold == 5
    http = ProbeSpec.parse({"httpGet": {"port": 80, "path": "/healthz"}})
    assert http.kind == "http" and http.path == "/healthz"
    ex = ProbeSpec.parse({"exec": {"command": ["/bin/true"]}})
    assert ex.kind == "exec" and ex.command == ["/bin/true"]
    assert ProbeSpec.parse(None) is None
    assert ProbeSpec.parse({}) is None
    assert ProbeSpec.parse({"grpc": {"port": 1}}) is None  # unsupported kind


def test_tcp_probe_against_real_socket():
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port = srv.getsockname()[1]
    try:
        assert run_probe(ProbeSpec(kind="tcp", port=port), {})
    finally:
        srv.close()
    assert not run_probe(ProbeSpec(kind="tcp", port=port), {})  # closed now


def test_exec_probe():
    assert run_probe(ProbeSpec(kind="exec", command=["/bin/true"]), {})
    assert not run_probe(ProbeSpec(kind="exec", command=["/bin/false"]), {})
    assert not run_probe(
        ProbeSpec(kind="exec", command=["/bin/sleep", "5"], timeout_s=0.2), {})


def test_http_probe():
    from http.server import BaseHTTPRequestHandler, HTTPServer

    class H(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_GET(self):
            code = 200 if self.path == "/ok" else 503
            self.send_response(code)
            self.send_header("Content-Length", "0")
            self.end_headers()

    srv = HTTPServer(("127.0.0.1", 0), H)
    port = srv.server_address[1]
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        assert run_probe(ProbeSpec(kind="http", port=port, path="/ok"), {})
        assert not run_probe(ProbeSpec(kind="http", port=port, path="/bad"), {})
    finally:
        srv.shutdown()


def test_thresholds():
    spec = ProbeSpec(kind="tcp", port=1, failure_threshold=3,
                     success_threshold=2)
    st = ProbeState()
    assert advance(st, spec, False) is None
    assert advance(st, spec, False) is None
    assert advance(st, spec, False) is False   # 3rd consecutive failure
    assert advance(st, spec, True) is False    # still unhealthy (needs 2)
    assert advance(st, spec, True) is True     # 2nd consecutive success
    assert advance(st, spec, False) is True    # 1 failure < threshold


# ---- e2e through the full stack ----

from k8s_runpod_kubelet_amd.app import build_stack
from k8s_runpod_kubelet_amd.config import Config
from k8s_runpod_kubelet_amd.kube.client import NotFoundError
from k8s_runpod_kubelet_amd.kube.fake import FakeKube
from tests.conftest import make_pod, wait_until


@pytest.fixture
def stack(tmp_state_dir):
    cfg = Config(state_dir=tmp_state_dir, gpu_count_override=8,
                 pending_retry_interval_s=0.2, notify_interval_s=0)
    kube = FakeKube()
    s = build_stack(cfg, client=kube)
    s.runtime.enable_cgroups = False
    s.start(serve_http=False)
    yield s, kube
    s.stop()


def _conds(kube, name):
    try:
        p = kube.get_pod("default", name)
    except NotFoundError:
        return None, {}
    return p, {c["type"]: c["status"]
               for c in p.get("status", {}).get("conditions", [])}


def test_readiness_probe_gates_ready(stack):
    """A tcpSocket readinessProbe owns Ready: the pod only becomes Ready
    once the port actually accepts, regardless of the process being up."""
    s, kube = stack
    from tests.conftest import free_port

    port = free_port()
    pod = make_pod("probed", command=["podworker"],
                   args=["--startup-delay", "1.5",
                         "--listen-port", str(port), "--hold"])
    pod["spec"]["containers"][0]["readinessProbe"] = {
        "tcpSocket": {"port": port}, "periodSeconds": 1,
        "failureThreshold": 1, "successThreshold": 1,
    }
    kube.create_pod("default", pod)
    time.sleep(0.8)  # process up, port not yet bound (startup delay)
    _, conds = _conds(kube, "probed")
    assert conds.get("Ready") != "True"

    def ready():
        p, c = _conds(kube, "probed")
        return p if c.get("Ready") == "True" else None

    assert wait_until(ready, timeout_s=20) is not None
    kube.delete_pod("default", "probed")


def test_liveness_probe_kills_and_restarts(stack):
    """A failing livenessProbe kills the container; restartPolicy=OnFailure
    brings it back (restartCount grows)."""
    s, kube = stack
    pod = make_pod("livefail", command=["podworker"], args=["--hold"])
    pod["spec"]["restartPolicy"] = "OnFailure"
    pod["spec"]["containers"][0]["livenessProbe"] = {
        "exec": {"command": ["/bin/false"]},
        "periodSeconds": 1, "failureThreshold": 2, "initialDelaySeconds": 0,
    }
    kube.create_pod("default", pod)

    def restarted():
        try:
            p = kube.get_pod("default", "livefail")
        except NotFoundError:
            return None
        css = p.get("status", {}).get("containerStatuses", [])
        return p if css and css[0].get("restartCount", 0) >= 1 else None

    assert wait_until(restarted, timeout_s=30) is not None
    kube.delete_pod("default", "livefail")
    assert wait_until(lambda: not s.ledger.reservations, timeout_s=15)


def test_startup_probe_gates_readiness_then_passes(process_runtime,
                                                   tmp_path):
    """startupProbe semantics: the container is not Ready (and liveness
    does not run) until the startup probe passes; with no readinessProbe,
    startup success makes it Ready."""
    import time

    from k8s_runpod_kubelet_amd.runtime.probes import ProbeSpec
    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerSpec, DeployParams, PodStatus)

    rt = process_runtime
    gate = tmp_path / "started-marker"
    st = rt.deploy(DeployParams(
        pod_key="default-sp1", name="sp1",
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"], args=["-c", "sleep 30"],
            startup=ProbeSpec(kind="exec",
                              command=["/usr/bin/test", "-f", str(gate)],
                              period_s=1.0, timeout_s=5.0,
                              failure_threshold=30))],
    ))
    time.sleep(2.2)  # a couple of startup attempts fail (marker absent)
    s = rt.get_detailed_status(st.id)
    assert not s.containers[0].ready, "ready before startup probe passed"
    gate.write_text("up")
    deadline = time.time() + 10
    while time.time() < deadline:
        s = rt.get_detailed_status(st.id)
        if s.containers[0].ready:
            break
        time.sleep(0.1)
    assert s.containers[0].ready, "startup pass did not mark Ready"
    rt.terminate(st.id)


def test_startup_probe_failure_kills_container(process_runtime):
    """failureThreshold exhausted on the startupProbe kills the container
    into the restartPolicy machinery (Never here -> pod fails)."""
    import time

    from k8s_runpod_kubelet_amd.runtime.probes import ProbeSpec
    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerSpec, DeployParams, PodStatus)

    rt = process_runtime
    st = rt.deploy(DeployParams(
        pod_key="default-sp2", name="sp2",
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"], args=["-c", "sleep 30"],
            startup=ProbeSpec(kind="exec", command=["/bin/false"],
                              period_s=1.0, timeout_s=5.0,
                              failure_threshold=2))],
    ))
    deadline = time.time() + 15
    while time.time() < deadline:
        s = rt.get_detailed_status(st.id)
        if s.desired_status == PodStatus.EXITED:
            break
        time.sleep(0.1)
    assert s.desired_status == PodStatus.EXITED
    assert s.containers[0].message == "startup probe failed"


def test_pre_stop_hook_runs_before_sigterm(process_runtime, tmp_path):
    """preStop lifecycle hook runs inside the grace window before SIGTERM
    reaches the container."""
    from k8s_runpod_kubelet_amd.runtime.probes import ProbeSpec
    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerSpec, DeployParams, PodStatus)
    import time

    rt = process_runtime
    marker = tmp_path / "prestop-ran"
    st = rt.deploy(DeployParams(
        pod_key="default-ps", name="ps", termination_grace_s=8.0,
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"],
            args=["-c", f"trap 'echo got-term; [ -f {marker} ] && "
                        f"echo HOOK-FIRST; exit 0' TERM; "
                        f"while :; do sleep 0.05; done"],
            pre_stop=ProbeSpec.parse_hook(
                {"exec": {"command": ["/bin/sh", "-c",
                                      f"touch {marker}"]}}))],
    ))
    time.sleep(0.3)
    rt.terminate(st.id)
    deadline = time.time() + 10
    while time.time() < deadline:
        s = rt.get_detailed_status(st.id)
        if s.desired_status == PodStatus.TERMINATED:
            break
        time.sleep(0.05)
    assert s.desired_status == PodStatus.TERMINATED
    assert marker.exists(), "preStop hook never ran"
    out = rt.get_logs(st.id)
    assert "HOOK-FIRST" in out, out  # marker existed when TERM arrived


def test_post_start_hook_failure_kills(process_runtime):
    """postStart failure kills the container (k8s semantics)."""
    from k8s_runpod_kubelet_amd.runtime.probes import ProbeSpec
    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerSpec, DeployParams, PodStatus)
    import time

    rt = process_runtime
    st = rt.deploy(DeployParams(
        pod_key="default-psf", name="psf",
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"], args=["-c", "sleep 30"],
            post_start=ProbeSpec.parse_hook(
                {"exec": {"command": ["/bin/false"]}}))],
    ))
    deadline = time.time() + 10
    while time.time() < deadline:
        s = rt.get_detailed_status(st.id)
        if s.desired_status == PodStatus.EXITED:
            break
        time.sleep(0.05)
    assert s.desired_status == PodStatus.EXITED
    assert s.containers[0].message == "postStart hook failed"


def test_post_start_sleep_hook_ok(process_runtime):
    """postStart sleep handler (k8s 1.29 sleep action) succeeds and leaves
    the container running."""
    from k8s_runpod_kubelet_amd.runtime.probes import ProbeSpec
    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerSpec, DeployParams, PodStatus)
    import time

    rt = process_runtime
    st = rt.deploy(DeployParams(
        pod_key="default-pss", name="pss",
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"], args=["-c", "sleep 30"],
            post_start=ProbeSpec.parse_hook({"sleep": {"seconds": 0.2}}))],
    ))
    time.sleep(0.8)
    s = rt.get_detailed_status(st.id)
    assert s.desired_status == PodStatus.RUNNING
    assert s.containers[0].exit_code is None
    rt.terminate(st.id)
