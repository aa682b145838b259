#!/usr/bin/env python3
"""Torch-free fast validation on a GPU box (small gpurun budgets): one GPU
podworker pod through the full stack, then the round-2-late features — an
OCI image pod pulled on-miss from a loopback registry, running read-only
rootfs as non-root, plus the runAsNonRoot refusal path. Prints PASS lines;
exits nonzero on any failure. Also runs on CPU (GPU pod section skipped)."""

import os
import subprocess
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, os.getcwd())

APP_C = r"""
#include <stdio.h>
#include <unistd.h>
int main(void) {
    printf("uid=%d\n", (int)getuid());
    FILE* w = fopen("/rootfs-write", "w");
    printf("rootfs=%s\n", w ? "writable" : "readonly");
    if (w) fclose(w);
    FILE* v = fopen("/scratch/f", "w");
    printf("volume=%s\n", v ? "writable" : "readonly");
    if (v) fclose(v);
    FILE* r = fopen("/etc/app-release", "r");
    printf("image-file=%s\n", r ? "present" : "missing");
    if (r) fclose(r);
    fflush(stdout);
    return 0;
}
"""


def main() -> int:
    t0 = time.time()
    from k8s_runpod_kubelet_amd.ops import load_native

    native = load_native(build_if_missing=False)
    gpus = native.enumerate_gpus("/sys")
    print(f"[{time.time()-t0:.1f}s] native loaded, {len(gpus)} GPUs")

    from k8s_runpod_kubelet_amd.app import build_stack
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.client import NotFoundError
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube

    state = tempfile.mkdtemp(prefix="amdvk-qc-")
    cfg = Config(state_dir=state, notify_interval_s=0.0,
                 pending_retry_interval_s=0.2,
                 gpu_count_override=(-1 if gpus else 2))
    kube = FakeKube()
    stack = build_stack(cfg, client=kube)
    stack.start(serve_http=False)
    failures = []
    try:
        if gpus:
            pod = {
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "qc-gpu", "namespace": "default"},
                "spec": {
                    "nodeName": cfg.node_name,
                    "containers": [{
                        "name": "main", "image": "amdvk/podworker:qc",
                        "command": ["podworker"],
                        "args": ["--expect-gpus", "1", "--hold"],
                        "resources": {"limits": {"amd.com/gpu": "1"}},
                    }],
                },
            }
            kube.create_pod("default", pod)
            deadline = time.time() + 60
            ready = False
            while time.time() < deadline:
                p = kube.get_pod("default", "qc-gpu")
                conds = (p.get("status") or {}).get("conditions") or []
                if any(c["type"] == "Ready" and c["status"] == "True"
                       for c in conds):
                    ready = True
                    break
                time.sleep(0.05)
            if ready:
                print(f"PASS gpu-pod-ready {time.time()-t0:.1f}s")
            else:
                failures.append("gpu pod never Ready")
            kube.delete_pod("default", "qc-gpu", grace_period_s=1)
            deadline = time.time() + 30
            while time.time() < deadline:
                try:
                    kube.get_pod("default", "qc-gpu")
                except NotFoundError:
                    break
                time.sleep(0.05)
            else:
                failures.append("gpu pod never finalized")
            if not failures:
                print(f"PASS gpu-pod-finalized {time.time()-t0:.1f}s")
        else:
            print("SKIP gpu-pod (no GPU)")

        # ---- image path: pull-on-miss + readOnlyRootFilesystem +
        # runAsNonRoot, through the provider (translate included) ----
        # Env probe (same as tests/test_gpu.py): in chroot fallback mode
        # (no CAP_SYS_ADMIN) image pods need device nodes outside /dev to
        # be openable; path-based LSM sandboxes (gpurun boxes) deny that
        # (open -> EACCES), which no runtime design can work around.
        import stat as statmod

        rt = stack.runtime
        run_image_pod = True
        if (rt._rootfs_mgr is not None
                and rt._rootfs_mgr.mode() == "chroot"):
            probe = Path(tempfile.mkdtemp(prefix="amdvk-qc-dev-")) / "null"
            try:
                st_null = os.stat("/dev/null")
                os.mknod(probe, st_null.st_mode, st_null.st_rdev)
                fd = os.open(probe, os.O_RDWR)
                os.close(fd)
            except (OSError, PermissionError):
                run_image_pod = False
                print("SKIP img-* (chroot mode + LSM denies device nodes "
                      "outside /dev — sandbox policy)")
            finally:
                if probe.exists():
                    probe.unlink()
        from k8s_runpod_kubelet_amd.runtime.oci import (ImageStore,
                                                        build_layout)
        from k8s_runpod_kubelet_amd.runtime.registry_server import (
            RegistryServer,
        )

        work = Path(tempfile.mkdtemp(prefix="amdvk-qc-img-"))
        (work / "app.c").write_text(APP_C)
        subprocess.run(["gcc", "-static", "-O1", "-o", str(work / "app"),
                        str(work / "app.c")], check=True)
        tree = work / "tree"
        (tree / "bin").mkdir(parents=True)
        (tree / "etc").mkdir()
        (tree / "scratch").mkdir()
        (work / "app").rename(tree / "bin" / "app")
        (tree / "etc" / "app-release").write_text("qc\n")
        (tree / "etc" / "passwd").write_text(
            "root:x:0:0::/:/bin/app\nqc:x:1500:1500::/:/bin/app\n")
        (tree / "etc" / "group").write_text("root:x:0:\nqc:x:1500:\n")
        layout = work / "layout"
        layout.mkdir()
        build_layout(str(layout), "qc/app:v1", str(tree),
                     entrypoint=["/bin/app"])
        remote = ImageStore(str(work / "remote"))
        remote.add_layout(str(layout), "qc/app:v1")
        srv = RegistryServer(remote).start()

        rt.image_registry = srv.url  # pull-on-miss into the empty store

        if not run_image_pod:
            # still validate the pull path itself (no rootfs needed)
            img = rt._pull_image("qc/app:v1")
            if img is not None:
                print("PASS img-pulled (client only)")
            else:
                failures.append("registry pull failed")
            srv.stop()
            if failures:
                print("FAILURES:", *failures, sep="\n  ")
                return 1
            print(f"ALL PASS (img-run env-skipped) in {time.time()-t0:.1f}s")
            return 0

        pod = {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "qc-img", "namespace": "default"},
            "spec": {
                "nodeName": cfg.node_name,
                "restartPolicy": "Never",
                "volumes": [{"name": "scratch", "emptyDir": {}}],
                "containers": [{
                    "name": "main", "image": "qc/app:v1",
                    "securityContext": {"runAsUser": 1500,
                                        "runAsGroup": 1500,
                                        "runAsNonRoot": True,
                                        "readOnlyRootFilesystem": True},
                    "volumeMounts": [{"name": "scratch",
                                      "mountPath": "/scratch"}],
                }],
            },
        }
        kube.create_pod("default", pod)
        deadline = time.time() + 60
        logs = ""
        while time.time() < deadline:
            p = kube.get_pod("default", "qc-img")
            phase = (p.get("status") or {}).get("phase")
            if phase in ("Succeeded", "Failed"):
                break
            time.sleep(0.05)
        else:
            failures.append("image pod never completed")
            phase = "timeout"
        prov = stack.provider
        logs = prov.get_container_logs("default", "qc-img", "main", 100)
        chroot_mode = (rt._rootfs_mgr is not None
                       and rt._rootfs_mgr.mode() == "chroot")
        checks = [("phase", phase == "Succeeded"),
                  ("pulled", rt.image_store.resolve("qc/app:v1")
                   is not None),
                  ("uid", "uid=1500" in logs),
                  ("rootfs-ro", "rootfs=readonly" in logs or chroot_mode),
                  ("volume-rw", "volume=writable" in logs),
                  ("image-file", "image-file=present" in logs)]
        for name, ok in checks:
            if ok:
                print(f"PASS img-{name}")
            else:
                failures.append(f"img-{name}: logs={logs!r} phase={phase}")

        # runAsNonRoot refusal: same image, no runAsUser -> image user root
        pod2 = {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "qc-nonroot", "namespace": "default"},
            "spec": {
                "nodeName": cfg.node_name, "restartPolicy": "Never",
                "containers": [{
                    "name": "main", "image": "qc/app:v1",
                    "securityContext": {"runAsNonRoot": True},
                }],
            },
        }
        kube.create_pod("default", pod2)
        deadline = time.time() + 20
        refused = False
        while time.time() < deadline:
            # the refusal surfaces as a Warning/DeployError event and the
            # pod stays Pending (CreateContainerConfigError analogue)
            with kube._lock:
                evs = list(kube.events.objects.values())
            if any("runAsNonRoot" in (e.get("message") or "")
                   and e.get("reason") == "DeployError" for e in evs):
                refused = True
                break
            time.sleep(0.05)
        if refused:
            print("PASS img-nonroot-refused")
        else:
            failures.append("runAsNonRoot violation not surfaced")
        srv.stop()
    finally:
        stack.stop()

    if failures:
        print("FAILURES:", *failures, sep="\n  ")
        return 1
    print(f"ALL PASS in {time.time()-t0:.1f}s")
    return 0


if __name__ == "__main__":
    sys.exit(main())
