"""Per-pod container rootfs preparation on top of the OCI store.

Two isolation modes, picked once per runtime from a capability probe:

- **mountns** (full): per-container overlayfs (lowerdir = shared unpacked
  image, upperdir/workdir private) + bind mounts for devices and host GPU
  userspace, mounted by the launcher child inside its own mount namespace,
  then pivot_root. Copy-free and the container's writes survive restarts in
  its upper layer.
- **chroot** (degraded, e.g. sandboxes without CAP_SYS_ADMIN): the image is
  copied into a private per-container rootfs, device nodes are mknod'd, and
  the child plain-chroots. No host mounts are possible, so GPU pods in this
  mode must carry ROCm userspace in the image — and the host must permit
  opening device nodes outside /dev (path-based LSM policies return EACCES
  there; measured on the gpurun sandbox, tests/test_gpu.py skip guard).

The launcher is fail-closed either way (launcher.cpp: a failed rootfs setup
never execs on the host), so a mode downgrade is an explicit decision here,
never an accident in the child."""

from __future__ import annotations

import logging
import os
import shutil
from dataclasses import dataclass, field
from pathlib import Path
from typing import List, Optional, Tuple

from . import mnt
from .oci import ImageStore, ResolvedImage

log = logging.getLogger("runtime.rootfs")

# devices every container gets (bound in mountns mode, mknod'd in chroot)
_BASE_DEVICES = ["null", "zero", "full", "random", "urandom", "tty"]


def _container_rel(path: str) -> str:
    """Normalize a container-absolute path from the POD SPEC (mountPath,
    workingDir) to a rootfs-relative path, clamping `..` at the root —
    a spec-controlled '/../../etc/x' must never address anything outside
    the per-container rootfs (in chroot mode that would be a host write)."""
    parts: list = []
    for p in path.split("/"):
        if p in ("", "."):
            continue
        if p == "..":
            if parts:
                parts.pop()
        else:
            parts.append(p)
    return "/".join(parts)


@dataclass
class PreparedRootfs:
    rootfs: str                   # path handed to the launcher
    mounts: List[tuple] = field(default_factory=list)
    chroot_only: bool = False
    container_dir: str = ""       # per-container state dir (for cleanup)


class RootfsManager:
    def __init__(self, store: ImageStore, containers_dir: str, native,
                 gpu_binds: Optional[List[str]] = None,
                 extra_binds: Optional[List[str]] = None,
                 isolation: str = "auto"):
        self.store = store
        self.containers_dir = Path(containers_dir)
        self.containers_dir.mkdir(parents=True, exist_ok=True)
        self._native = native
        # host paths bound read-only into GPU pods (driver userspace — the
        # thin-image-plus-host-driver pattern GPU container runtimes use)
        self.gpu_binds = list(gpu_binds or [])
        self.extra_binds = list(extra_binds or [])
        self.isolation = isolation
        self._mode: Optional[str] = None

    def mode(self) -> str:
        if self._mode is None:
            if self.isolation in ("mountns", "chroot"):
                self._mode = self.isolation
            else:
                self._mode = ("mountns" if self._native.probe_mount_namespace()
                              else "chroot")
                log.info("image isolation mode selected",
                         extra={"mode": self._mode})
        return self._mode

    def downgrade_to_chroot(self) -> None:
        """Called when mountns launches fail at runtime (e.g. overlayfs
        missing): all subsequent image pods use the chroot mode."""
        if self._mode != "chroot":
            log.warning("downgrading image isolation to chroot mode")
            self._mode = "chroot"

    # ---- preparation ----

    def prepare(self, instance_id: str, container_name: str,
                image: ResolvedImage, hostname: str,
                gpu_device_paths: Optional[List[str]] = None,
                working_dir: str = "",
                volume_binds: Optional[List[tuple]] = None,
                host_aliases: Optional[List[tuple]] = None,
                read_only: bool = False,
                resolv_conf: str = "") -> PreparedRootfs:
        """volume_binds: (host_src, container_dst, ro) tuples from the
        pod's volumes (emptyDir/hostPath/secret/configMap projections,
        materialized by the runtime). read_only = securityContext.
        readOnlyRootFilesystem: the overlay mountpoint is remounted
        MS_RDONLY (volume/tmpfs sub-mounts stay writable — k8s
        semantics); unenforceable in chroot mode (warned, degraded)."""
        cdir = self.containers_dir / f"{instance_id}-{container_name}"
        aliases = list(host_aliases or [])
        if self.mode() == "mountns":
            return self._prepare_overlay(cdir, image, hostname,
                                         gpu_device_paths or [], working_dir,
                                         volume_binds or [], aliases,
                                         read_only, resolv_conf)
        if read_only:
            log.warning("readOnlyRootFilesystem not enforceable in chroot "
                        "fallback mode; continuing writable")
        return self._prepare_chroot(cdir, image, hostname,
                                    gpu_device_paths or [], working_dir,
                                    volume_binds or [], aliases, resolv_conf)

    def _parse_bind(self, entry: str) -> Tuple[str, str, bool]:
        """'src[:dst[:ro|rw]]' -> (src, dst, ro). Default dst=src, ro."""
        parts = entry.split(":")
        src = parts[0]
        dst = parts[1] if len(parts) > 1 and parts[1] else src
        ro = (parts[2] if len(parts) > 2 else "ro") != "rw"
        return src, dst, ro

    def _prepare_overlay(self, cdir: Path, image: ResolvedImage,
                         hostname: str, gpu_devices: List[str],
                         working_dir: str, volume_binds: List[tuple],
                         host_aliases: List[tuple],
                         read_only: bool = False,
                         resolv_conf: str = "") -> PreparedRootfs:
        lower = self.store.rootfs_for(image)
        upper = cdir / "upper"
        work = cdir / "work"
        merged = cdir / "merged"
        for d in (upper, work, merged):
            d.mkdir(parents=True, exist_ok=True)
        # pre-create every mount target in the upper layer (the child only
        # issues mount(2); plain upper dirs/files merge, they don't shadow)
        for d in (".amdvk-oldroot", "proc", "sys", "dev", "dev/shm",
                  "dev/dri", "tmp", "etc"):
            (upper / d).mkdir(parents=True, exist_ok=True)
        for f in _BASE_DEVICES:
            (upper / "dev" / f).touch()
        if working_dir:
            (upper / _container_rel(working_dir)).mkdir(parents=True,
                                                        exist_ok=True)
        self._write_identity(upper, hostname, host_aliases, resolv_conf)

        mounts = [mnt.overlay(str(merged), str(lower), str(upper),
                              str(work), ro=read_only),
                  mnt.proc(str(merged / "proc")),
                  # ro sysfs view (ROCm userspace reads KFD topology from
                  # /sys/class/kfd; standard for non-netns containers)
                  mnt.bind("/sys", str(merged / "sys"), ro=True),
                  mnt.tmpfs(str(merged / "dev" / "shm"))]
        for f in _BASE_DEVICES:
            mounts.append(mnt.bind(f"/dev/{f}", str(merged / "dev" / f)))
        bind_entries = list(self.extra_binds)
        if gpu_devices:
            bind_entries += self.gpu_binds
        for dev in gpu_devices:
            rel = dev.lstrip("/")
            target = upper / rel
            target.parent.mkdir(parents=True, exist_ok=True)
            target.touch()
            mounts.append(mnt.bind(dev, str(merged / rel)))
        for entry in bind_entries:
            src, dst, ro = self._parse_bind(entry)
            if not os.path.exists(src):
                continue
            rel = _container_rel(dst)
            target = upper / rel
            if os.path.isdir(src):
                target.mkdir(parents=True, exist_ok=True)
            else:
                target.parent.mkdir(parents=True, exist_ok=True)
                target.touch()
            mounts.append(mnt.bind(src, str(merged / rel), ro=ro))
        for src, dst, ro, kind in volume_binds:
            rel = _container_rel(dst)
            target = upper / rel
            if kind == "tmpfs":
                # emptyDir.medium=Memory: src carries sizeLimit bytes
                target.mkdir(parents=True, exist_ok=True)
                opts = "mode=1777"
                if src and src != "0":
                    opts += f",size={src}"
                mounts.append(mnt.tmpfs(str(merged / rel), opts))
                continue
            if os.path.isdir(src):
                target.mkdir(parents=True, exist_ok=True)
            else:
                target.parent.mkdir(parents=True, exist_ok=True)
                target.touch()
            mounts.append(mnt.bind(src, str(merged / rel), ro=ro))
        return PreparedRootfs(rootfs=str(merged), mounts=mounts,
                              chroot_only=False, container_dir=str(cdir))

    def _prepare_chroot(self, cdir: Path, image: ResolvedImage,
                        hostname: str, gpu_devices: List[str],
                        working_dir: str, volume_binds: List[tuple],
                        host_aliases: List[tuple],
                        resolv_conf: str = "") -> PreparedRootfs:
        rootfs = cdir / "rootfs"
        if not (rootfs / ".amdvk-ready").exists():
            cdir.mkdir(parents=True, exist_ok=True)
            cache = self.store.rootfs_for(image)
            if rootfs.exists():
                shutil.rmtree(rootfs)
            shutil.copytree(cache, rootfs, symlinks=True)
            for d in ("proc", "sys", "dev", "dev/shm", "tmp", "etc"):
                (rootfs / d).mkdir(parents=True, exist_ok=True)
            for f in _BASE_DEVICES:
                self._mknod_like(f"/dev/{f}", rootfs / "dev" / f)
            for dev in gpu_devices:
                rel = dev.lstrip("/")
                (rootfs / rel).parent.mkdir(parents=True, exist_ok=True)
                self._mknod_like(dev, rootfs / rel)
            if working_dir:
                (rootfs / _container_rel(working_dir)).mkdir(parents=True,
                                                             exist_ok=True)
            self._write_identity(rootfs, hostname, host_aliases,
                                 resolv_conf)
            # chroot mode has no mounts: volumes are materialized INTO the
            # per-container copy (per-container — cross-container emptyDir
            # sharing needs the mount-namespace mode; hostPath cannot be
            # provided at all without bind mounts)
            for src, dst, ro, kind in volume_binds:
                if kind == "hostPath":
                    shutil.rmtree(cdir, ignore_errors=True)
                    raise RuntimeError(
                        "hostPath volumes require mount-namespace "
                        "isolation (unavailable here)")
                rel = _container_rel(dst)
                target = rootfs / rel
                if os.path.isdir(src):
                    shutil.copytree(src, target, dirs_exist_ok=True)
                else:
                    target.parent.mkdir(parents=True, exist_ok=True)
                    shutil.copy2(src, target)
            (rootfs / ".amdvk-ready").touch()
        return PreparedRootfs(rootfs=str(rootfs), mounts=[],
                              chroot_only=True, container_dir=str(cdir))

    @staticmethod
    def _mknod_like(host_dev: str, target: Path) -> None:
        """Replicate a host device node (major/minor) into the rootfs —
        the chroot-mode substitute for a bind mount. Best-effort: skipped
        without CAP_MKNOD."""
        try:
            st = os.stat(host_dev)
        except OSError:
            return
        try:
            target.unlink(missing_ok=True)
            os.mknod(target, st.st_mode, st.st_rdev)
        except (OSError, PermissionError):
            log.debug("mknod unavailable", extra={"dev": host_dev})

    def _write_identity(self, root: Path, hostname: str,
                        host_aliases: List[tuple],
                        resolv_conf: str = "") -> None:
        """kubelet-managed identity files (k8s writes these per pod),
        including spec.hostAliases lines and resolv.conf (dnsPolicy
        Default = the node's resolver; spec.dnsConfig overrides)."""
        etc = root / "etc"
        etc.mkdir(parents=True, exist_ok=True)
        lines = ["127.0.0.1\tlocalhost",
                 f"127.0.1.1\t{hostname or 'pod'}"]
        for ip, names in host_aliases:
            lines.append(f"{ip}\t" + " ".join(names))
        if not resolv_conf:
            try:
                resolv_conf = Path("/etc/resolv.conf").read_text()
            except OSError:
                resolv_conf = ""
        try:
            (etc / "hostname").write_text((hostname or "pod") + "\n")
            (etc / "hosts").write_text("\n".join(lines) + "\n")
            if resolv_conf:
                (etc / "resolv.conf").write_text(resolv_conf)
        except OSError:
            pass

    # ---- exec path resolution inside the image ----

    def resolve_argv0(self, image: ResolvedImage, prepared: PreparedRootfs,
                      argv0: str) -> str:
        """PATH-resolve the entrypoint inside the image (execve does no
        PATH search; the merged view isn't mounted in the parent, so search
        the backing tree)."""
        if "/" in argv0:
            return argv0
        backing = (Path(prepared.rootfs) if prepared.chroot_only
                   else self.store.rootfs_for(image))
        for p in image.config.path_env.split(":"):
            candidate = backing / p.lstrip("/") / argv0
            if candidate.exists():
                return "/" + str(candidate.relative_to(backing))
        return argv0  # let exec fail with a truthful ENOENT

    def cleanup(self, instance_id: str) -> None:
        for d in self.containers_dir.glob(f"{instance_id}-*"):
            shutil.rmtree(d, ignore_errors=True)
