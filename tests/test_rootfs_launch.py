"""Native launcher rootfs execution: mount namespace + pivot_root (and the
degraded chroot mode). These exercise _native.launch_process directly with
a hand-built rootfs containing a static binary — the mechanism under the
image-backed runtime (reference contract: the image actually runs,
runpod_client.go:1304)."""

import os
import subprocess
import time
from pathlib import Path

import pytest

from k8s_runpod_kubelet_amd.ops import load_native
from k8s_runpod_kubelet_amd.runtime import mnt

native = load_native()

pytestmark = pytest.mark.skipif(
    os.geteuid() != 0, reason="rootfs launch tests need root")

HELLO_C = r"""
#include <stdio.h>
#include <unistd.h>
#include <stdlib.h>
int main(int argc, char** argv) {
    FILE* f = fopen("/etc/marker.txt", "r");
    if (!f) { printf("no-marker\n"); } else {
        char buf[256] = {0};
        fread(buf, 1, sizeof(buf) - 1, f);
        printf("marker=%s\n", buf);
        fclose(f);
    }
    // host leak check: this exists on the host, must not in the container
    printf("host-python=%s\n",
           access("/usr/bin/python3", F_OK) == 0 ? "visible" : "absent");
    printf("pid=%d\n", (int)getpid());
    printf("uid=%d\n", (int)getuid());
    char* probe = getenv("AMDVK_TEST_WRITE");
    if (probe) {
        FILE* w = fopen(probe, "w");
        if (w) { fputs("written", w); fclose(w); printf("wrote=ok\n"); }
        else printf("wrote=fail\n");
    }
    if (argc > 1 && argv[1][0] == 's') sleep(30);
    fflush(stdout);
    return 0;
}
"""


@pytest.fixture(scope="module")
def hello_bin(tmp_path_factory):
    d = tmp_path_factory.mktemp("hello")
    src = d / "hello.c"
    src.write_text(HELLO_C)
    out = d / "hello"
    subprocess.run(["gcc", "-static", "-O1", "-o", str(out), str(src)],
                   check=True)
    return out


def make_rootfs(tmp_path, hello_bin, name="rootfs"):
    rootfs = tmp_path / name
    (rootfs / "bin").mkdir(parents=True)
    (rootfs / "etc").mkdir()
    (rootfs / "proc").mkdir()
    (rootfs / "tmp").mkdir()
    (rootfs / ".amdvk-oldroot").mkdir()
    import shutil

    shutil.copy2(hello_bin, rootfs / "bin" / "hello")
    (rootfs / "etc" / "marker.txt").write_text("in-container")
    return rootfs


def run_and_wait(pid, pidfd, timeout_s=10.0):
    loop = native.EventLoop()
    loop.add_process(pid, pidfd, -1, 0)
    deadline = time.time() + timeout_s
    while time.time() < deadline:
        for ev in loop.poll(100):
            if ev.type == "exited":
                return ev.exit_code
    native.signal_process(pid, 9, True)
    raise AssertionError("child did not exit")


def test_pivot_root_isolation(tmp_path, hello_bin):
    if not native.probe_mount_namespace():
        pytest.skip("no mount-namespace capability here")
    rootfs = make_rootfs(tmp_path, hello_bin)
    log = tmp_path / "out.log"
    mounts = [mnt.bind(str(rootfs), str(rootfs)),
              mnt.proc(str(rootfs / "proc")),
              mnt.tmpfs(str(rootfs / "tmp"))]
    pid, pidfd, _, _, _ = native.launch_process(
        ["/bin/hello"], ["AMDVK_TEST_WRITE=/tmp/w.txt"],
        "", str(log), str(log), "", True, False, -1, -1,
        True, "podhost", str(rootfs), False, mounts)
    rc = run_and_wait(pid, pidfd)
    out = log.read_text()
    assert rc == 0, out
    assert "marker=in-container" in out      # image content visible
    assert "host-python=absent" in out       # host filesystem NOT visible
    assert "pid=1\n" in out                  # own PID namespace
    assert "wrote=ok" in out                 # tmpfs writable
    # the tmpfs write happened in the container's mount ns, not on the host
    assert not (rootfs / "tmp" / "w.txt").exists()


def test_chroot_fallback(tmp_path, hello_bin):
    rootfs = make_rootfs(tmp_path, hello_bin, "rootfs2")
    log = tmp_path / "out2.log"
    pid, pidfd, _, _, _ = native.launch_process(
        ["/bin/hello"], ["AMDVK_TEST_WRITE=/tmp/w.txt"],
        "", str(log), str(log), "", True, False, -1, -1,
        False, "", str(rootfs), True, [])
    rc = run_and_wait(pid, pidfd)
    out = log.read_text()
    assert rc == 0, out
    assert "marker=in-container" in out
    assert "host-python=absent" in out
    assert "wrote=ok" in out
    # chroot mode: the write lands in the per-pod rootfs copy (no tmpfs)
    assert (rootfs / "tmp" / "w.txt").read_text() == "written"


def test_rootfs_with_credential_drop(tmp_path, hello_bin):
    if not native.probe_mount_namespace():
        pytest.skip("no mount-namespace capability here")
    rootfs = make_rootfs(tmp_path, hello_bin, "rootfs3")
    os.chmod(rootfs / "bin" / "hello", 0o755)
    log = tmp_path / "out3.log"
    mounts = [mnt.bind(str(rootfs), str(rootfs))]
    pid, pidfd, _, _, _ = native.launch_process(
        ["/bin/hello"], [], "", str(log), str(log), "", True, False,
        65534, 65534, False, "", str(rootfs), False, mounts)
    rc = run_and_wait(pid, pidfd)
    out = log.read_text()
    assert rc == 0, out
    assert "uid=65534" in out
    assert "host-python=absent" in out


def test_failed_mount_never_execs_on_host(tmp_path, hello_bin):
    """A broken mount plan must abort the launch (125), never run the
    entrypoint against the host filesystem."""
    if not native.probe_mount_namespace():
        pytest.skip("no mount-namespace capability here")
    rootfs = make_rootfs(tmp_path, hello_bin, "rootfs4")
    bad = [("/nonexistent-src-xyz", str(rootfs / "etc"), "", "",
            mnt.MS_BIND, False)]
    with pytest.raises(RuntimeError, match="rootfs setup"):
        native.launch_process(
            ["/bin/hello"], [], "", "", "", "", True, False, -1, -1,
            False, "", str(rootfs), False,
            [mnt.bind(str(rootfs), str(rootfs))] + bad)


def test_readonly_bind(tmp_path, hello_bin):
    if not native.probe_mount_namespace():
        pytest.skip("no mount-namespace capability here")
    rootfs = make_rootfs(tmp_path, hello_bin, "rootfs5")
    ro_src = tmp_path / "ro-src"
    ro_src.mkdir()
    (ro_src / "data.txt").write_text("ro")
    (rootfs / "mnt").mkdir()
    log = tmp_path / "out5.log"
    mounts = [mnt.bind(str(rootfs), str(rootfs)),
              mnt.bind(str(ro_src), str(rootfs / "mnt"), ro=True)]
    pid, pidfd, _, _, _ = native.launch_process(
        ["/bin/hello"], ["AMDVK_TEST_WRITE=/mnt/should-fail"],
        "", str(log), str(log), "", True, False, -1, -1,
        False, "", str(rootfs), False, mounts)
    rc = run_and_wait(pid, pidfd)
    out = log.read_text()
    assert rc == 0, out
    assert "wrote=fail" in out               # read-only bind enforced
    assert not (ro_src / "should-fail").exists()
