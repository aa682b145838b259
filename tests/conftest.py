import os
import sys
import tempfile

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X GPU (run via gpurun)"
    )


def pytest_sessionfinish(session, exitstatus):
    """ROCm's C++ static destructors can std::terminate at interpreter exit
    (a joinable library thread torn down out of order), turning a fully
    green GPU run into SIGABRT/rc=134 — observed intermittently on MI355X
    boxes AFTER '14 passed' was printed. When HIP was initialized in this
    process, install the exit guard (flush + drain earlier atexit hooks +
    _exit) with pytest's own status. Test results are unaffected: this runs
    only after the session (and its reporting) is complete, propagates the
    real exit status, and — unlike a bare os._exit — still runs any
    instrumentation hooks the harness registered at process start
    (utils/exit_guard.py)."""
    try:
        import torch

        # is_available() alone: the abort comes from ROCm libraries loaded
        # into this process, whether or not a CUDA context was created here
        # (is_initialized() can stay False while library threads exist).
        gpu_touched = torch.cuda.is_available()
    except Exception:
        gpu_touched = False
    if gpu_touched:
        from k8s_runpod_kubelet_amd.utils.exit_guard import install

        install(int(exitstatus))


@pytest.fixture
def tmp_state_dir():
    with tempfile.TemporaryDirectory(prefix="amdvk-test-") as d:
        yield d


@pytest.fixture
def fake_kube():
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube

    return FakeKube()


@pytest.fixture
def synthetic_ledger():
    from k8s_runpod_kubelet_amd.gpu.inventory import Inventory
    from k8s_runpod_kubelet_amd.gpu.ledger import Ledger

    inv = Inventory(synthetic_count=8)
    inv.discover()
    ledger = Ledger(inv)
    ledger.sync_inventory()
    return ledger


@pytest.fixture
def process_runtime(synthetic_ledger, tmp_state_dir):
    from k8s_runpod_kubelet_amd.runtime.process_runtime import ProcessRuntime

    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    yield rt
    drain_runtime(rt)
    rt.close()


def make_pod(name="p1", namespace="default", node="virtual-runpod", gpus=0,
             annotations=None, command=None, args=None, ports=None,
             containers=None, labels=None, owner=None,
             restart_policy="Never"):
    """Test pod factory (dict in wire shape). restart_policy defaults to
    Never because most tests run pods to completion (as a user would for a
    Job-style pod); pass None to omit the field and exercise apiserver/
    kubelet defaulting (k8s default: Always)."""
    if containers is None:
        c = {"name": "main", "image": "amdvk/test:latest"}
        if command is not None:
            c["command"] = command
        if args is not None:
            c["args"] = args
        if gpus:
            c["resources"] = {"limits": {"amd.com/gpu": str(gpus)}}
        if ports:
            c["ports"] = [{"containerPort": p, "protocol": "TCP"} for p in ports]
        containers = [c]
    pod = {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {"name": name, "namespace": namespace,
                     "annotations": dict(annotations or {}),
                     "labels": dict(labels or {})},
        "spec": {"nodeName": node, "containers": containers},
    }
    if restart_policy is not None:
        pod["spec"]["restartPolicy"] = restart_policy
    if owner:
        pod["metadata"]["ownerReferences"] = [owner]
    return pod


@pytest.fixture
def pod_factory():
    return make_pod


def wait_until(fn, timeout_s=5.0, interval_s=0.01):
    import time

    deadline = time.time() + timeout_s
    while time.time() < deadline:
        result = fn()
        if result:
            return result
        time.sleep(interval_s)
    return fn()


@pytest.fixture
def waiter():
    return wait_until


_port_counter = [0]


@pytest.fixture(scope="session", autouse=True)
def _reap_leaked_pod_processes():
    """Session-end sweep: kill pod processes left by kubelet-crash-
    semantics teardowns (stack.stop() deliberately leaves pods running
    for adoption). Matches ONLY this repo's podworker binary (by exe
    path) and the rotation test's writer (by exact cmdline) — never a
    name pattern."""
    yield
    import signal as _signal

    if os.environ.get("PYTEST_XDIST_WORKER"):
        # under xdist this fixture ends per-WORKER while sibling workers
        # still run live pods — sweeping here killed them mid-test.
        # Parallel runs rely on the per-fixture drains; only serial runs
        # (and the xdist controller-less case) sweep.
        return

    from k8s_runpod_kubelet_amd.ops import podworker_binary

    try:
        target = os.path.realpath(podworker_binary())
    except Exception:
        return
    me = os.getpid()
    for pid in os.listdir("/proc"):
        if not pid.isdigit() or int(pid) == me:
            continue
        try:
            exe = os.readlink(f"/proc/{pid}/exe")
        except OSError:
            continue
        kill = exe in (target, target + " (deleted)")
        if not kill:
            try:
                with open(f"/proc/{pid}/cmdline", "rb") as fh:
                    cmd = fh.read().replace(b"\0", b" ").decode(
                        errors="replace")
                kill = ("printf 'xxxxxxxxxxxxxxxxxxxxxxxxxxxxxxxx%.0s'"
                        in cmd)
            except OSError:
                continue
        if kill:
            try:
                os.kill(int(pid), _signal.SIGKILL)
            except OSError:
                pass


def drain_runtime(rt, timeout_s: float = 5.0) -> None:
    """Kill every live instance (grace 0) and wait for the processes to
    die. Test teardown MUST do this before close(): close() cancels kill
    timers (kubelet-crash semantics — the next kubelet re-arms them), so
    a bare close() after terminate() leaks TERM-immune pid-1 workloads
    (this once leaked 58 infinite log writers totalling ~115 GB of
    deleted-but-open files)."""
    import time as _time

    try:
        for inst in rt.list_instances():
            try:
                rt.terminate(inst.id, grace_override_s=0.0)
            except Exception:
                pass
        deadline = _time.time() + timeout_s
        while _time.time() < deadline:
            alive = [i for i in rt.list_instances()
                     if any(c.exit_code is None for c in i.containers)]
            if not alive:
                break
            _time.sleep(0.05)
    except Exception:
        pass


def free_port() -> int:
    """A free TCP port from a per-xdist-worker range. A plain
    bind(0)+close is racy across parallel workers (another worker can
    grab the port before the test's process binds it — seen as a
    port-gated-readiness flake); disjoint per-worker ranges remove the
    cross-worker race, and the bind probe skips anything else living
    in the range."""
    import os
    import socket

    worker = os.environ.get("PYTEST_XDIST_WORKER", "gw0")
    try:
        widx = int(worker.lstrip("gw") or 0)
    except ValueError:
        widx = 0
    base = 21000 + 1000 * (widx % 40)
    for _ in range(1000):
        _port_counter[0] += 1
        port = base + (_port_counter[0] % 1000)
        probe = socket.socket()
        try:
            probe.bind(("127.0.0.1", port))
        except OSError:
            continue
        finally:
            probe.close()
        return port
    raise RuntimeError("no free port in worker range")
