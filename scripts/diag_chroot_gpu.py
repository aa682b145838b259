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


def snapshot(src, tree, max_files=5000):
    """Bounded, symlink-free sysfs snapshot (sysfs symlinks form cycles —
    never follow them) of small text attributes."""
    if not os.path.isdir(src):
        print(f"  [snapshot] missing {src}")
        return
    # resolve the top-level dir itself (e.g. /sys/class/kfd -> devices path)
    # but never follow links below it
    count = 0
    for root, _dirs, files in os.walk(src, followlinks=False):
        rel = os.path.relpath(root, "/")
        os.makedirs(tree / rel, exist_ok=True)
        for f in files:
            fp = os.path.join(root, f)
            if os.path.islink(fp):
                continue
            try:
                if os.path.getsize(fp) > 1 << 20:
                    continue
                with open(fp, "rb") as fh:
                    data = fh.read(1 << 20)
            except OSError:
                continue
            (tree / rel / f).write_bytes(data)
            count += 1
            if count >= max_files:
                print(f"  [snapshot] {src}: truncated at {max_files} files")
                return


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

    # sysfs snapshots HSA/libdrm are known to read. /sys/class/kfd is a
    # symlink to devices/virtual — walk the real path, store at BOTH names.
    snapshot("/sys/devices/virtual/kfd/kfd/topology", tree)
    real = tree / "sys/devices/virtual/kfd"
    cls = tree / "sys/class/kfd"
    if real.is_dir() and not cls.exists():
        cls.parent.mkdir(parents=True, exist_ok=True)
        cls.symlink_to("../devices/virtual/kfd")
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
        # /sys/dev/char/<maj:min> -> device dir: record the link target
        # shallowly (one level of attributes), not the whole PCI tree
        sysdev = f"/sys/dev/char/{maj}:{minr}"
        if os.path.exists(sysdev):
            realdev = os.path.realpath(sysdev)
            reldev = os.path.relpath(realdev, "/")
            os.makedirs(tree / reldev, exist_ok=True)
            for f in os.listdir(realdev):
                fp = os.path.join(realdev, f)
                if os.path.islink(fp) or os.path.isdir(fp):
                    continue
                try:
                    (tree / reldev / f).write_bytes(open(fp, "rb").read())
                except OSError:
                    pass
            link = tree / "sys/dev/char" / f"{maj}:{minr}"
            link.parent.mkdir(parents=True, exist_ok=True)
            if not link.exists():
                # container-relative target (an absolute host path would
                # resolve wrongly inside the chroot)
                link.symlink_to(os.path.relpath(realdev, "/sys/dev/char"))
    # /proc files read-only consumers want (plain files in the rootfs)
    for pf in ("/proc/cpuinfo", "/proc/meminfo", "/proc/version"):
        try:
            (tree / pf.lstrip("/")).write_bytes(open(pf, "rb").read())
        except OSError:
            pass

    # LD_PRELOAD syscall spy: log every open/openat/stat failure to stderr
    # so we can see exactly which path ROCr/ROCt misses inside the chroot
    spy_c = td / "spy.c"
    spy_c.write_text(r"""
#define _GNU_SOURCE
#include <stdarg.h>
#include <stdio.h>
#include <dlfcn.h>
#include <errno.h>
#include <fcntl.h>
static int (*real_openat)(int, const char*, int, ...) = 0;
static int (*real_open)(const char*, int, ...) = 0;
int openat(int dirfd, const char* path, int flags, ...) {
    if (!real_openat) real_openat = dlsym(RTLD_NEXT, "openat");
    va_list ap; va_start(ap, flags);
    int mode = va_arg(ap, int); va_end(ap);
    int rc = real_openat(dirfd, path, flags, mode);
    if (rc < 0) fprintf(stderr, "[spy] openat(%s) -> %d\n", path, -errno);
    return rc;
}
int open(const char* path, int flags, ...) {
    if (!real_open) real_open = dlsym(RTLD_NEXT, "open");
    va_list ap; va_start(ap, flags);
    int mode = va_arg(ap, int); va_end(ap);
    int rc = real_open(path, flags, mode);
    if (rc < 0) fprintf(stderr, "[spy] open(%s) -> %d\n", path, -errno);
    return rc;
}
""")
    subprocess.run(["gcc", "-shared", "-fPIC", "-O1", "-o",
                    str(tree / "spy.so"), str(spy_c), "-ldl"], check=True)

    log = td / "out.log"
    env = {
        "PATH": "/bin",
        "LD_LIBRARY_PATH": "/opt/rocm/lib:/opt/rocm/lib64",
        "LD_PRELOAD": "/spy.so",
        "AMD_LOG_LEVEL": "4",
        "HSAKMT_DEBUG_LEVEL": "7",
        "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        "ROCR_VISIBLE_DEVICES": "0",
    }
    pid, pidfd, _, _, _ = native.launch_process(
        ["/bin/podworker", "--expect-gpus", "1", "--run-for", "0.2"],
        [f"{k}={v}" for k, v in env.items()],
        "", str(log), str(log), "", True, False, -1, -1,
        False, "", str(tree), True, [])
    loop = native.EventLoop()
    loop.add_process(pid, pidfd, -1, 0)
    deadline = time.time() + 60
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
