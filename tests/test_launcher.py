"""Native launcher + event loop tests (CPU-only)."""

import os
import time

import pytest

from k8s_runpod_kubelet_amd.ops import load_native


@pytest.fixture(scope="module")
def native():
    return load_native()


def base_env():
    return [f"{k}={v}" for k, v in os.environ.items()]


def drain(loop, want, timeout_s=5.0):
    events = []
    deadline = time.time() + timeout_s
    while time.time() < deadline:
        events += loop.poll(100)
        if any(e.type == want for e in events):
            return events
    return events


def test_launch_exit_code(native, tmp_path):
    loop = native.EventLoop()
    pid, pidfd, ready_fd, _spawn_s, _cg_s = native.launch_process(
        ["/bin/sh", "-c", "exit 7"], base_env(),
        "", str(tmp_path / "out.log"), "", "", True, True)
    loop.add_process(pid, pidfd, ready_fd, 1)
    events = drain(loop, "exited")
    exited = [e for e in events if e.type == "exited"]
    assert exited and exited[0].exit_code == 7
    assert loop.tracked_count() == 0


def test_ready_pipe_protocol(native, tmp_path):
    loop = native.EventLoop()
    pid, pidfd, ready_fd, _spawn_s, _cg_s = native.launch_process(
        ["/bin/bash", "-c", 'echo READY >&$AMDVK_READY_FD; sleep 0.1'],
        base_env(), "", str(tmp_path / "out.log"), "", "", True, True)
    loop.add_process(pid, pidfd, ready_fd, 2)
    events = drain(loop, "exited")
    types = [e.type for e in events]
    assert "ready" in types
    ready = next(e for e in events if e.type == "ready")
    assert b"READY" in ready.data if isinstance(ready.data, bytes) else "READY" in ready.data


def test_exec_failure_reported(native, tmp_path):
    with pytest.raises(RuntimeError, match="spawn"):
        native.launch_process(
            ["/no/such/binary"], base_env(), "", "", "", "", True, True)


def test_signal_process(native, tmp_path):
    loop = native.EventLoop()
    pid, pidfd, ready_fd, _spawn_s, _cg_s = native.launch_process(
        ["/bin/sleep", "30"], base_env(), "", str(tmp_path / "o.log"), "", "",
        True, True)
    loop.add_process(pid, pidfd, ready_fd, 3)
    assert native.signal_process(pid, 15, True) == 0  # whole group (setsid)
    events = drain(loop, "exited")
    exited = [e for e in events if e.type == "exited"]
    assert exited and exited[0].exit_code == 128 + 15


def test_stdout_redirect(native, tmp_path):
    log = tmp_path / "redir.log"
    loop = native.EventLoop()
    pid, pidfd, ready_fd, _spawn_s, _cg_s = native.launch_process(
        ["/bin/sh", "-c", "echo hello-out; echo hello-err >&2"],
        base_env(), "", str(log), "", "", True, True)
    loop.add_process(pid, pidfd, ready_fd, 4)
    drain(loop, "exited")
    text = log.read_text()
    assert "hello-out" in text and "hello-err" in text


def test_open_pidfd_adoption(native):
    pid, pidfd, ready_fd, _spawn_s, _cg_s = native.launch_process(
        ["/bin/sleep", "0.2"], base_env(), "", "", "", "", True, False)
    os.close(pidfd)
    # Re-open (adoption path) and watch the exit through a fresh loop.
    pidfd2 = native.open_pidfd(pid)
    assert pidfd2 >= 0
    loop = native.EventLoop()
    loop.add_process(pid, pidfd2, -1, 9)
    events = drain(loop, "exited")
    assert any(e.type == "exited" and e.exit_code == 0 for e in events)
    assert native.open_pidfd(2**22 - 1) < 0  # nonexistent pid


def test_cgroup_helpers_best_effort(native, tmp_path):
    # Against a plain directory: mkdir works, controller writes fail.
    path = str(tmp_path / "cg" / "pod1")
    native.cgroup_create(path, "", "")
    assert os.path.isdir(path)
    assert native.cgroup_proc_count(path) == -1  # no cgroup.procs file
    assert native.cgroup_remove(path)


@pytest.mark.parametrize("sanitizer", ["thread", "address,undefined"])
def test_native_sanitizer_stress(sanitizer, tmp_path):
    """Build the launcher + event loop with TSan/ASan+UBSan and hammer them
    from 8 threads (spawn + mid-flight removal + concurrent polling). The
    reference ships no race detection at all and has known data races
    (SURVEY §5.2); the native hot path here must be clean under both."""
    import shutil
    import subprocess

    if shutil.which("g++") is None:
        pytest.skip("no g++ on this box")
    csrc = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "k8s_runpod_kubelet_amd", "ops", "csrc")
    binary = tmp_path / f"stress_{sanitizer.split(',')[0]}"
    build = subprocess.run(
        ["g++", "-O1", "-g", f"-fsanitize={sanitizer}", "-std=c++17",
         os.path.join(csrc, "launcher.cpp"),
         os.path.join(csrc, "stress_launcher.cpp"),
         "-o", str(binary), "-lpthread"],
        capture_output=True, text=True, timeout=120,
    )
    assert build.returncode == 0, build.stderr[-2000:]
    run = subprocess.run(
        [str(binary), "8", "25"], capture_output=True, text=True, timeout=120,
        env={"PATH": "/usr/bin:/bin", "TSAN_OPTIONS": "halt_on_error=1",
             "ASAN_OPTIONS": "detect_leaks=0"},
    )
    assert run.returncode == 0, (run.stdout + run.stderr)[-4000:]
    assert run.stdout.startswith("ok:"), run.stdout


def test_clone3_born_in_cgroup(native, tmp_path):
    """clone3(CLONE_INTO_CGROUP) fast path: the child's first instruction
    already runs inside the pod cgroup — no cgroup.procs migration (measured
    ~130 ms under churn on MI355X, profiles/). Verified by reading the
    child's /proc/<pid>/cgroup."""
    import secrets

    cgroot = None
    for cand in ("/sys/fs/cgroup", "/sys/fs/cgroup/unified"):
        if os.path.exists(os.path.join(cand, "cgroup.procs")) and \
                os.access(cand, os.W_OK):
            cgroot = cand
            break
    if cgroot is None:
        pytest.skip("no writable cgroup2 hierarchy on this box")
    cgdir = os.path.join(cgroot, f"amdvk-test-{secrets.token_hex(4)}")
    try:
        os.mkdir(cgdir)
    except OSError:
        pytest.skip("cgroup2 root not writable")
    try:
        out = tmp_path / "out.log"
        pid, pidfd, ready_fd, spawn_s, cgroup_s = native.launch_process(
            ["/bin/bash", "-c",
             "echo READY >&${AMDVK_READY_FD}; sleep 5"],
            base_env(), "", str(out), str(out), cgdir, True, True,
        )
        try:
            with open(f"/proc/{pid}/cgroup") as fh:
                line = fh.read().strip()
            # cgroup2 entry: "0::<path>"; the path must be our pod cgroup.
            assert line.splitlines()[-1].startswith("0::"), line
            assert line.rsplit("::", 1)[-1].endswith(os.path.basename(cgdir)), line
            assert cgroup_s == 0.0  # born attached: nothing was migrated
            loop = native.EventLoop()
            loop.add_process(pid, pidfd, ready_fd, pid)
            events = drain(loop, "ready")
            assert any(e.type == "ready" for e in events)
        finally:
            native.signal_process(pid, 9, True)
            time.sleep(0.1)
    finally:
        for _ in range(50):
            try:
                os.rmdir(cgdir)
                break
            except OSError:
                time.sleep(0.05)


def test_cgroup_device_filter_enforced(native, tmp_path):
    """The hand-assembled BPF_PROG_TYPE_CGROUP_DEVICE program actually
    enforces: inside the filtered cgroup, major 1 (mem devices) is denied
    except minor 5 (/dev/zero) — so /dev/zero opens and /dev/null fails
    with EPERM. (In production the denied major is 226/DRM: only the bound
    GPUs' render nodes are reachable.)"""
    import secrets
    import subprocess
    import sys

    cgroot = None
    for cand in ("/sys/fs/cgroup", "/sys/fs/cgroup/unified"):
        if os.path.exists(os.path.join(cand, "cgroup.procs")) and \
                os.access(cand, os.W_OK):
            cgroot = cand
            break
    if cgroot is None:
        pytest.skip("no writable cgroup2 hierarchy")
    cgdir = os.path.join(cgroot, f"amdvk-dev-{secrets.token_hex(4)}")
    os.mkdir(cgdir)
    try:
        if not native.cgroup_attach_device_filter(cgdir, 1, [5]):
            pytest.skip("BPF device filter not permitted on this box")
        probe = (
            "import os\n"
            "os.close(os.open('/dev/zero', os.O_RDONLY))\n"  # allowed minor
            "try:\n"
            "    os.open('/dev/null', os.O_WRONLY)\n"
            "except PermissionError:\n"
            "    print('DENIED-OK'); raise SystemExit(0)\n"
            "print('NOT-DENIED'); raise SystemExit(1)\n"
        )
        out = tmp_path / "probe.log"
        pid, pidfd, _, _, _ = native.launch_process(
            [sys.executable, "-c", probe], base_env(), "",
            str(out), str(out), cgdir, True, False,
        )
        loop = native.EventLoop()
        loop.add_process(pid, pidfd, -1, pid)
        events = drain(loop, "exited", timeout_s=15)
        exit_ev = [e for e in events if e.type == "exited"]
        assert exit_ev and exit_ev[0].exit_code == 0, out.read_text()
        assert "DENIED-OK" in out.read_text()
    finally:
        for _ in range(50):
            try:
                os.rmdir(cgdir)
                break
            except OSError:
                time.sleep(0.05)


def test_run_as_user_drops_credentials(native, tmp_path):
    """securityContext.runAsUser: the clone3 child setgid/setuids before
    exec; the workload must observe the dropped identity."""
    if os.geteuid() != 0:
        pytest.skip("needs root to drop to another uid")
    out = tmp_path / "id.log"
    pid, pidfd, _, _, _ = native.launch_process(
        ["/usr/bin/id"], ["PATH=/usr/bin:/bin"], "",
        str(out), str(out), "", True, False, 65534, 65534,
    )
    loop = native.EventLoop()
    loop.add_process(pid, pidfd, -1, pid)
    events = drain(loop, "exited", timeout_s=10)
    assert any(e.type == "exited" and e.exit_code == 0 for e in events)
    text = out.read_text()
    assert "uid=65534" in text and "gid=65534" in text, text


def test_run_as_user_never_falls_back_to_root(native, tmp_path):
    """A pod that asked for dropped credentials must fail loudly rather
    than silently run as the kubelet's user when the fast path can't be
    used (argv[0] without '/': no PATH search on the execve path)."""
    with pytest.raises(RuntimeError, match="runAsUser"):
        native.launch_process(
            ["id"], ["PATH=/usr/bin:/bin"], "", "", "", "",
            True, False, 65534, -1,
        )


def test_pod_pid_uts_namespaces(native, tmp_path):
    """Container-like isolation: the pod process is pid 1 in its own PID
    namespace with its own hostname (UTS ns), and the parent's hostname is
    untouched. Skips where CAP_SYS_ADMIN is absent (the native path then
    degrades to no-namespace spawn automatically)."""
    import socket
    import subprocess
    import sys

    probe = ("import os,socket;"
             "print('NS', os.getpid(), socket.gethostname())")
    out = tmp_path / "ns.log"
    pid, pidfd, _, _, _ = native.launch_process(
        [sys.executable, "-c", probe], base_env(), "",
        str(out), str(out), "", True, False, -1, -1, True, "pod-xyz",
    )
    loop = native.EventLoop()
    loop.add_process(pid, pidfd, -1, pid)
    events = drain(loop, "exited", timeout_s=15)
    assert any(e.type == "exited" and e.exit_code == 0 for e in events)
    text = out.read_text()
    if "NS 1 pod-xyz" not in text:
        assert f"NS" in text, text
        pytest.skip(f"namespaces degraded on this box: {text!r}")
    assert socket.gethostname() != "pod-xyz"  # parent UTS untouched
