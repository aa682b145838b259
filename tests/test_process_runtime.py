"""ProcessRuntime lifecycle tests (CPU: podworker runs in no-GPU mode)."""

import time

import pytest

from k8s_runpod_kubelet_amd.runtime.process_runtime import ProcessRuntime
from k8s_runpod_kubelet_amd.runtime.types import (
    ContainerSpec,
    DeployParams,
    PodStatus,
    is_successful_completion,
)


def params(pod_key="default-p1", args=None, containers=None, **kw):
    if containers is None:
        containers = [ContainerSpec(name="main", command=["podworker"],
                                    args=args or ["--hold"])]
    return DeployParams(pod_key=pod_key, name=pod_key.split("-", 1)[1],
                        containers=containers, **kw)


def wait_status(rt, iid, status, timeout=5.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        s = rt.get_detailed_status(iid)
        if s.desired_status == status:
            return s
        time.sleep(0.01)
    return rt.get_detailed_status(iid)


def wait_ready(rt, iid, timeout=5.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        s = rt.get_detailed_status(iid)
        if s.containers and all(c.ready for c in s.containers):
            return s
        time.sleep(0.005)
    return rt.get_detailed_status(iid)


def test_deploy_ready_exit(process_runtime):
    rt = process_runtime
    st = rt.deploy(params(args=["--run-for", "0.2"]))
    assert st.desired_status == PodStatus.RUNNING
    s = wait_ready(rt, st.id)
    assert s.containers[0].ready
    s = wait_status(rt, st.id, PodStatus.EXITED)
    assert s.desired_status == PodStatus.EXITED
    assert s.exit_code == 0
    assert is_successful_completion(s)


def test_failure_exit_code(process_runtime):
    rt = process_runtime
    st = rt.deploy(params(args=["--exit-code", "3", "--run-for", "0.05"]))
    s = wait_status(rt, st.id, PodStatus.EXITED)
    assert s.exit_code == 3
    assert not is_successful_completion(s)


def test_terminate_sigterm(process_runtime):
    rt = process_runtime
    st = rt.deploy(params(args=["--hold"]))
    wait_ready(rt, st.id)
    rt.terminate(st.id)
    s = wait_status(rt, st.id, PodStatus.TERMINATED)
    assert s.desired_status == PodStatus.TERMINATED


def test_logs(process_runtime):
    rt = process_runtime
    st = rt.deploy(params(args=["--run-for", "0.05"]))
    wait_status(rt, st.id, PodStatus.EXITED)
    logs = rt.get_logs(st.id)
    assert "podworker: ready" in logs
    assert rt.get_logs("nonexistent") == ""


def test_gpu_reservation_lifecycle(synthetic_ledger, tmp_state_dir):
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    try:
        st = rt.deploy(params(gpu_count=2, gpu_memory_bytes=64 * 1024**3,
                              args=["--run-for", "0.2"]))
        assert len(st.gpu_indices) == 2
        res = synthetic_ledger.get_reservation("default-p1")
        assert res is not None and res.bytes_per_gpu == 32 * 1024**3
        wait_status(rt, st.id, PodStatus.EXITED)
        # HBM returns to the ledger on exit, before pod deletion
        assert synthetic_ledger.get_reservation("default-p1") is None
    finally:
        rt.close()


def test_subscriber_push(process_runtime):
    events = []
    process_runtime.subscribe(lambda iid: events.append(iid))
    st = process_runtime.deploy(params(args=["--run-for", "0.05"]))
    wait_status(process_runtime, st.id, PodStatus.EXITED)
    # the exit notification is delivered by the event thread slightly
    # after the status flips — wait, don't sleep a fixed beat
    from tests.conftest import wait_until

    assert wait_until(lambda: events.count(st.id) >= 2, timeout_s=5), \
        events  # ready + exit


def test_not_found(process_runtime):
    assert process_runtime.get_detailed_status("nope").desired_status == PodStatus.NOT_FOUND
    assert process_runtime.get_status("nope") == PodStatus.NOT_FOUND


def test_multi_container_pod(process_runtime):
    rt = process_runtime
    p = params(containers=[
        ContainerSpec(name="a", command=["podworker"], args=["--run-for", "0.1"]),
        ContainerSpec(name="b", command=["podworker"], args=["--run-for", "0.3"]),
    ])
    st = rt.deploy(p)
    assert len(st.containers) == 2
    # EXITED only after *all* containers exit
    s = wait_status(rt, st.id, PodStatus.EXITED, timeout=5)
    assert s.desired_status == PodStatus.EXITED
    assert {c.name for c in s.containers} == {"a", "b"}
    assert rt.get_logs(st.id, "b")


def test_persistence_and_adoption(synthetic_ledger, tmp_state_dir):
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    st = rt.deploy(params(gpu_count=1, args=["--hold"]))
    wait_ready(rt, st.id)
    # Simulate kubelet restart: stop watcher, drop state, re-adopt from disk.
    rt._stop.set()
    rt._loop.wake()
    rt._watcher.join(timeout=2)

    from k8s_runpod_kubelet_amd.gpu.inventory import Inventory
    from k8s_runpod_kubelet_amd.gpu.ledger import Ledger

    inv2 = Inventory(synthetic_count=8)
    inv2.discover()
    ledger2 = Ledger(inv2)
    ledger2.sync_inventory()
    rt2 = ProcessRuntime(ledger2, tmp_state_dir, enable_cgroups=False)
    try:
        adopted = rt2.adopt_persisted()
        assert st.id in adopted
        s = rt2.get_detailed_status(st.id)
        assert s.desired_status == PodStatus.RUNNING
        assert ledger2.get_reservation("default-p1") is not None  # no double-bind
        rt2.terminate(st.id)
        s = wait_status(rt2, st.id, PodStatus.TERMINATED)
        assert s.desired_status == PodStatus.TERMINATED
    finally:
        rt2.close()
        rt.close()


def test_adoption_of_vanished_process(synthetic_ledger, tmp_state_dir):
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    st = rt.deploy(params(args=["--run-for", "30"]))
    wait_ready(rt, st.id)
    # Kill the process outside the runtime AND stop the watcher first so the
    # exit is never observed (simulates dying while the kubelet is down).
    rt._stop.set(); rt._loop.wake(); rt._watcher.join(timeout=2)
    import os, signal
    pid = rt.get_detailed_status(st.id).containers[0].pid
    os.kill(pid, signal.SIGKILL)
    os.waitpid(pid, 0)  # reap: in a real restart the process is fully gone

    rt2 = ProcessRuntime(synthetic_ledger.__class__(rt.ledger.inventory), tmp_state_dir,
                         enable_cgroups=False)
    # note: fresh ledger instance over same inventory
    rt2.ledger.sync_inventory()
    try:
        rt2.adopt_persisted()
        s = rt2.get_detailed_status(st.id)
        assert s.desired_status == PodStatus.EXITED
        assert s.containers[0].exit_code == -1  # vanished marker
    finally:
        rt2.close()
        rt.close()


def test_termination_grace_period(process_runtime):
    """spec.terminationGracePeriodSeconds: a TERM-ignoring workload is
    SIGKILLed after the pod's own grace window, not the global default."""
    import time

    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerSpec, DeployParams, PodStatus)

    rt = process_runtime
    st = rt.deploy(DeployParams(
        pod_key="default-stubborn", name="stubborn",
        termination_grace_s=0.5,
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"],
            args=["-c", "trap '' TERM; sleep 60"])],
    ))
    time.sleep(0.2)  # shell up with TERM ignored
    t0 = time.time()
    rt.terminate(st.id)
    deadline = time.time() + 10
    while time.time() < deadline:
        s = rt.get_detailed_status(st.id)
        if s.desired_status == PodStatus.TERMINATED:
            break
        time.sleep(0.02)
    assert s.desired_status == PodStatus.TERMINATED
    took = time.time() - t0
    assert 0.4 < took < 5, took  # killed at ~0.5 s grace, not 10 s
    assert s.containers[0].exit_code == 128 + 9


def test_adoption_preserves_credentials_and_deadline(synthetic_ledger,
                                                     tmp_state_dir):
    """Round-1 advisory (high): securityContext credentials, workingDir,
    activeDeadlineSeconds and terminationGracePeriodSeconds must survive a
    kubelet restart — a crash-restarting container was being relaunched as
    root with default grace/deadline."""
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    st = rt.deploy(DeployParams(
        pod_key="default-cred", name="cred",
        restart_policy="Always",
        active_deadline_s=3600.0,
        termination_grace_s=2.5,
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"], args=["-c", "sleep 60"],
            run_as_uid=12345, run_as_gid=54321, working_dir="/tmp")],
        init_containers=[ContainerSpec(
            name="setup", command=["/bin/true"],
            run_as_uid=12345, run_as_gid=54321, working_dir="/tmp")],
    ))
    wait_status(rt, st.id, PodStatus.RUNNING)
    rt._stop.set(); rt._loop.wake(); rt._watcher.join(timeout=2)

    rt2 = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    try:
        rt2.adopt_persisted()
        with rt2._lock:
            inst = rt2._instances[st.id]
        p = inst.params
        assert p.active_deadline_s == 3600.0
        assert p.termination_grace_s == 2.5
        c = p.containers[0]
        assert (c.run_as_uid, c.run_as_gid, c.working_dir) == (
            12345, 54321, "/tmp")
        ic = p.init_containers[0]
        assert (ic.run_as_uid, ic.run_as_gid, ic.working_dir) == (
            12345, 54321, "/tmp")
        # the re-armed deadline timer exists for the adopted RUNNING pod
        assert st.id in rt2._deadline_timers
        rt2.terminate(st.id)
        wait_status(rt2, st.id, PodStatus.TERMINATED)
    finally:
        rt2.close()
        rt.close()


def test_adoption_rearms_expired_deadline(synthetic_ledger, tmp_state_dir):
    """activeDeadlineSeconds whose budget expired while the kubelet was
    down fires immediately on adoption (remaining ≤ 0)."""
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    st = rt.deploy(params(pod_key="default-dl", args=["--hold"],
                          active_deadline_s=0.3))
    wait_status(rt, st.id, PodStatus.RUNNING)
    rt._stop.set(); rt._loop.wake(); rt._watcher.join(timeout=2)
    # cancel the first runtime's own deadline timer so IT doesn't kill the
    # process — the restart must enforce the deadline itself
    with rt._lock:
        t = rt._deadline_timers.pop(st.id, None)
    if t:
        t.cancel()
    time.sleep(0.4)  # budget now blown

    rt2 = ProcessRuntime(synthetic_ledger, tmp_state_dir, enable_cgroups=False)
    try:
        rt2.adopt_persisted()
        s = wait_status(rt2, st.id, PodStatus.EXITED, timeout=5)
        assert s.desired_status == PodStatus.EXITED
        assert s.last_error == "DeadlineExceeded"
    finally:
        rt2.close()
        rt.close()


def test_deadline_during_init_keeps_reason(process_runtime):
    """Round-1 advisory (low): a pod killed by activeDeadlineSeconds while
    an init container runs must still report DeadlineExceeded, not the init
    kill's exit message."""
    rt = process_runtime
    st = rt.deploy(DeployParams(
        pod_key="default-initdl", name="initdl",
        active_deadline_s=0.2,
        init_containers=[ContainerSpec(
            name="slow-init", command=["/bin/sh"], args=["-c", "sleep 30"])],
        containers=[ContainerSpec(
            name="main", command=["/bin/true"])],
    ))
    s = wait_status(rt, st.id, PodStatus.EXITED, timeout=5)
    assert s.desired_status == PodStatus.EXITED
    assert s.last_error == "DeadlineExceeded"


def test_exec_probe_runs_with_container_credentials(process_runtime):
    """Round-1 advisory (medium): exec probes must run with the container's
    runAsUser, not as the kubelet (root). The readiness probe here succeeds
    only when its uid is the container's 65534."""
    import os

    if os.geteuid() != 0:
        pytest.skip("needs root to exercise credential drop")
    rt = process_runtime
    from k8s_runpod_kubelet_amd.runtime.probes import ProbeSpec

    st = rt.deploy(DeployParams(
        pod_key="default-probeuid", name="probeuid",
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"], args=["-c", "sleep 30"],
            run_as_uid=65534, run_as_gid=65534,
            readiness=ProbeSpec(kind="exec",
                                command=["/bin/sh", "-c",
                                         '[ "$(id -u)" = 65534 ]'],
                                period_s=1.0, timeout_s=5.0))],
    ))
    s = wait_ready(rt, st.id, timeout=10.0)
    assert s.containers[0].ready, \
        "exec probe did not run as the container's uid"
    rt.terminate(st.id)
    wait_status(rt, st.id, PodStatus.TERMINATED)


def test_port_exposure_attributed_to_pod_pids(process_runtime):
    """Port-gated readiness must attribute listening sockets to the pod's
    own processes (socket inodes via /proc/<pid>/fd) — an unrelated process
    listening on the pod's port must NOT mark it exposed (round-1 weak #3;
    reference gates on per-instance portMappings, kubelet.go:566-605)."""
    import socket

    rt = process_runtime
    # unrelated listener (this test process) on port B
    other = socket.socket()
    other.bind(("127.0.0.1", 0))
    other.listen(1)
    port_b = other.getsockname()[1]
    # pick a second free port for the pod itself
    probe = socket.socket()
    probe.bind(("127.0.0.1", 0))
    port_a = probe.getsockname()[1]
    probe.close()
    try:
        st = rt.deploy(params(
            pod_key="default-ports",
            containers=[ContainerSpec(
                name="main", command=["podworker"],
                args=["--hold", "--listen-port", str(port_a)],
                tcp_ports=[port_a, port_b])],
        ))

        def pod_port_up():
            s = rt.get_detailed_status(st.id)
            return s if port_a in s.port_mappings else None

        deadline = time.time() + 10
        s = None
        while time.time() < deadline:
            s = pod_port_up()
            if s:
                break
            time.sleep(0.05)
        assert s is not None, "pod's own listener never detected"
        # the unrelated process's port must not count as exposed
        assert port_b not in s.port_mappings, \
            "host-wide socket leaked into the pod's port mappings"
        rt.terminate(st.id)
        wait_status(rt, st.id, PodStatus.TERMINATED)
    finally:
        other.close()


def test_partial_deploy_failure_kills_launched_containers(process_runtime):
    """Multi-container pod where the SECOND container fails to spawn: the
    already-running first container must be killed and untracked, not leak
    as an orphan process (found by adversarial review)."""
    rt = process_runtime
    with pytest.raises(Exception):
        rt.deploy(DeployParams(
            pod_key="default-partial", name="partial",
            containers=[
                ContainerSpec(name="good", command=["/bin/sh"],
                              args=["-c", "sleep 60"]),
                ContainerSpec(name="bad",
                              command=["/nonexistent-binary-xyz"]),
            ],
        ))
    # the good container's process must be gone and nothing tracked
    deadline = time.time() + 5
    while time.time() < deadline and rt.tracked_process_count() > 0:
        time.sleep(0.05)
    assert rt.tracked_process_count() == 0


def test_log_rotation_caps_chatty_containers(synthetic_ledger,
                                             tmp_state_dir):
    """A container writing unbounded output gets copytruncate-rotated at
    the cap (content preserved in the --previous slot; disk bounded)."""
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir,
                        enable_cgroups=False, log_max_bytes=64 * 1024)
    try:
        st = rt.deploy(DeployParams(
            pod_key="default-chatty", name="chatty",
            containers=[ContainerSpec(
                name="main", command=["/bin/sh"],
                args=["-c", "while :; do printf "
                            "'xxxxxxxxxxxxxxxxxxxxxxxxxxxxxxxx%.0s' "
                            "$(seq 200); sleep 0.02; done"])],
        ))
        from pathlib import Path

        logp = Path(tmp_state_dir) / "logs" / f"{st.id}-main.log"
        deadline = time.time() + 30
        rotated = False
        while time.time() < deadline:
            prev = Path(str(logp) + ".prev")
            if prev.exists() and prev.stat().st_size >= 64 * 1024:
                rotated = True
                break
            time.sleep(0.2)
        assert rotated, "log never rotated"
        # live file stays bounded (cap + a few seconds of writes)
        assert logp.stat().st_size < 2 * 64 * 1024 + 512 * 1024
        rt.terminate(st.id, grace_override_s=0.0)
        deadline = time.time() + 5
        while time.time() < deadline:
            if all(c.exit_code is not None
                   for c in rt.get_detailed_status(st.id).containers):
                break
            time.sleep(0.05)
    finally:
        from tests.conftest import drain_runtime

        drain_runtime(rt)
        rt.close()


def test_ephemeral_container_adoption(synthetic_ledger, tmp_state_dir):
    """A live ephemeral (kubectl-debug) container survives a kubelet
    restart: re-adopted, watched, and killed with the pod."""
    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir,
                        enable_cgroups=False)
    st = rt.deploy(params(pod_key="default-ephad", args=["--hold"]))
    wait_ready(rt, st.id)
    rt.add_ephemeral_container(st.id, ContainerSpec(
        name="dbg", command=["/bin/sh"], args=["-c", "sleep 60"]))
    time.sleep(0.2)
    eph_pid = rt.get_detailed_status(st.id).ephemeral_containers[0].pid
    assert eph_pid > 0
    rt._stop.set(); rt._loop.wake(); rt._watcher.join(timeout=2)

    rt2 = ProcessRuntime(synthetic_ledger, tmp_state_dir,
                         enable_cgroups=False)
    try:
        rt2.adopt_persisted()
        s = rt2.get_detailed_status(st.id)
        eph = s.ephemeral_containers[0]
        assert eph.name == "dbg" and eph.pid == eph_pid
        assert eph.exit_code is None
        import os as _os

        assert _os.path.exists(f"/proc/{eph_pid}")
        rt2.terminate(st.id)
        s = wait_status(rt2, st.id, PodStatus.TERMINATED)
        assert s.desired_status == PodStatus.TERMINATED
        deadline = time.time() + 5
        while time.time() < deadline and _os.path.exists(f"/proc/{eph_pid}"):
            time.sleep(0.05)
        assert not _os.path.exists(f"/proc/{eph_pid}"), \
            "ephemeral survived pod termination"
    finally:
        rt2.close()
        rt.close()


def test_adoption_rearms_kill_timer_for_terminating(synthetic_ledger,
                                                    tmp_state_dir):
    """Kubelet dies mid-grace while a TERM-immune pid-1 ignores SIGTERM:
    the next kubelet re-arms the SIGKILL timer on adoption — the
    container must not outlive the grace window forever."""
    import os

    rt = ProcessRuntime(synthetic_ledger, tmp_state_dir,
                        enable_cgroups=False)
    st = rt.deploy(DeployParams(
        pod_key="default-immune", name="immune",
        termination_grace_s=1.0,
        containers=[ContainerSpec(
            name="main", command=["/bin/sh"],
            args=["-c", "trap '' TERM; while :; do sleep 1; done"])],
    ))
    wait_status(rt, st.id, PodStatus.RUNNING)
    pid = rt.get_detailed_status(st.id).containers[0].pid
    rt.terminate(st.id)  # TERM ignored; grace timer armed
    rt.close()           # kubelet "dies": timer cancelled
    assert os.path.exists(f"/proc/{pid}")

    rt2 = ProcessRuntime(synthetic_ledger, tmp_state_dir,
                         enable_cgroups=False)
    try:
        rt2.adopt_persisted()
        deadline = time.time() + 10
        gone = False
        while time.time() < deadline:
            if not os.path.exists(f"/proc/{pid}") or \
                    open(f"/proc/{pid}/stat").read().rsplit(
                        ") ", 1)[-1].startswith("Z"):
                gone = True
                break
            time.sleep(0.1)
        assert gone, "TERM-immune container survived re-armed grace"
        s = wait_status(rt2, st.id, PodStatus.TERMINATED)
        assert s.desired_status == PodStatus.TERMINATED
    finally:
        from tests.conftest import drain_runtime

        drain_runtime(rt2)
        rt2.close()
