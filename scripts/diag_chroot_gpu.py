"""Diagnose chroot-mode GPU image pods: build a self-contained podworker
rootfs (library closure + sysfs/proc snapshots + mknod'd devices), chroot
into it with ROCm debug logging, and print what HSA complains about.
Run on a GPU box: python scripts/diag_chroot_gpu.py"""

import os
import re
import shutil
import stat
import subprocess
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from k8s_runpod_kubelet_amd.ops import load_native, podworker_binary  # noqa: E402

native = load_native()


def closure(binary):
    out = subprocess.run(["ldd", binary], capture_output=True, text=True,
                         check=True).stdout
    deps = {}
    for line in out.splitlines():
        m = re.search(r"=>\s+(\S+)\s+\(", line)
        if m:
            deps[m.group(1)] = m.group(1)
        else:
            m2 = re.search(r"^\s*(/\S*ld-linux\S*)\s+\(", line)
            if m2:
                deps[m2.group(1)] = m2.group(1)
    return deps


def snapshot(src, tree, follow=True):
    if not os.path.isdir(src):
        print(f"  [snapshot] missing {src}")
        return
    for root, _dirs, files in os.walk(src, followlinks=follow):
        rel = os.path.relpath(root, "/")
        os.makedirs(tree / rel, exist_ok=True)
        for f in files:
            try:
                with open(os.path.join(root, f), "rb") as fh:
                    data = fh.read()
            except OSError:
                continue
            (tree / rel / f).write_bytes(data)


def main():
    td = Path(tempfile.mkdtemp(prefix="diagchroot-"))
    tree = td / "rootfs"
    (tree / "bin").mkdir(parents=True)
    (tree / "dev" / "dri").mkdir(parents=True)
    (tree / "proc").mkdir()
    (tree / "tmp").mkdir()
    pw = podworker_binary()
    shutil.copy2(pw, tree / "bin" / "podworker")
    for cpath, hpath in closure(pw).items():
        dst = tree / cpath.lstrip("/")
        dst.parent.mkdir(parents=True, exist_ok=True)
        shutil.copy2(os.path.realpath(hpath), dst)
    real_rocm = os.path.realpath("/opt/rocm")
    if real_rocm != "/opt/rocm" and (tree / "opt").is_dir():
        link = tree / real_rocm.lstrip("/")
        if not link.exists():
            link.symlink_to("rocm")

    # sysfs snapshots HSA/libdrm are known to read
    snapshot("/sys/class/kfd/kfd/topology", tree)
    snapshot("/sys/devices/virtual/kfd/kfd/topology", tree)
    snapshot("/sys/devices/system/node", tree)
    # /sys/dev/char/<maj:min> device links for kfd + render nodes
    gpus = native.enumerate_gpus("/sys")
    devices = ["/dev/kfd"] + [f"/dev/dri/renderD{g.render_minor}"
                              for g in gpus if g.render_minor >= 0]
    for dev in devices:
        try:
            st = os.stat(dev)
        except OSError:
            print(f"  [dev] missing {dev}")
            continue
        rel = dev.lstrip("/")
        target = tree / rel
        target.parent.mkdir(parents=True, exist_ok=True)
        try:
            os.mknod(target, st.st_mode, st.st_rdev)
            print(f"  [dev] mknod {dev} ok "
                  f"({os.major(st.st_rdev)}:{os.minor(st.st_rdev)})")
        except OSError as exc:
            print(f"  [dev] mknod {dev} FAILED: {exc}")
        maj, minr = os.major(st.st_rdev), os.minor(st.st_rdev)
        sysdev = f"/sys/dev/char/{maj}:{minr}"
        if os.path.exists(sysdev):
            snapshot(os.path.realpath(sysdev), tree)
    # /proc files read-only consumers want (plain files in the rootfs)
    for pf in ("/proc/cpuinfo", "/proc/meminfo"):
        try:
            (tree / pf.lstrip("/")).write_bytes(open(pf, "rb").read())
        except OSError:
            pass

    log = td / "out.log"
    env = {
        "PATH": "/bin",
        "LD_LIBRARY_PATH": "/opt/rocm/lib:/opt/rocm/lib64",
        "AMD_LOG_LEVEL": "4",
        "HSAKMT_DEBUG_LEVEL": "7",
        "ROCR_VISIBLE_DEVICES": "0",
    }
    pid, pidfd, _, _, _ = native.launch_process(
        ["/bin/podworker", "--expect-gpus", "1", "--run-for", "0.2"],
        [f"{k}={v}" for k, v in env.items()],
        "", str(log), str(log), "", True, False, -1, -1,
        False, "", str(tree), True, [])
    loop = native.EventLoop()
    loop.add_process(pid, pidfd, -1, 0)
    deadline = time.time() + 120
    code = None
    while time.time() < deadline and code is None:
        for ev in loop.poll(200):
            if ev.type == "exited":
                code = ev.exit_code
    print(f"exit code: {code}")
    text = log.read_text(errors="replace")
    print("---- last 120 lines of pod log ----")
    print("\n".join(text.splitlines()[-120:]))


if __name__ == "__main__":
    main()
