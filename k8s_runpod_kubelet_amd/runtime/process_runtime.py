"""ProcessRuntime — local host-process pod runtime with event-driven status.

The MI355X-native data path that replaces the reference's RunPod cloud
(reference pkg/virtual_kubelet/runpod_client.go): a pod's containers run as
host processes launched by the native C++ launcher (ops/csrc/launcher.cpp)
with:

- GPU scoping via ROCR_VISIBLE_DEVICES/HIP_VISIBLE_DEVICES from the binder
  (the server-side GPU attach analogue),
- a per-pod cgroup v2 slice (cpu.max / memory.max) — best-effort when
  unprivileged,
- stdout/stderr captured to per-container log files (GetContainerLogs is real
  here, unlike the reference's "not supported" stub, kubelet.go:2027-2066),
- lifecycle *pushed* over pidfd+epoll: readiness from the AMDVK_READY_FD
  pipe, exit from the pidfd — so the provider learns of state changes in
  microseconds instead of on the next 10 s poll tick (kubelet.go:719).

"Image" handling (reference contract runpod_client.go:1304 — the image
actually runs): a pod whose image resolves in the node-local OCI store
(runtime/oci.py) executes the image's entrypoint inside a per-container
rootfs (runtime/rootfs.py: overlay + mount-ns + pivot_root, or chroot in
degraded sandboxes), with k8s command/args vs Entrypoint/Cmd semantics,
image Env/WorkingDir/User, and pod volumes bound in. Images absent from the
store keep the legacy host-binary behavior (the reserved `amdvk/*` family,
bench and test workloads use the in-tree HIP `podworker`); a `command`-less
unresolved-image container falls back to podworker in hold mode — mirroring
how the reference test pod runs a GPU-probe image (runpod_test.go:99).

Instance state is journaled under <state_dir>/instances/<id>.json so a
restarted kubelet re-adopts live pods (reference LoadRunning analogue,
kubelet.go:1380-1535) without double-binding GPUs.
"""

from __future__ import annotations

import dataclasses
import json
import logging
import os
import secrets
import shutil
import threading
import time
from pathlib import Path
from typing import Callable, Dict, List, Optional

from ..gpu.binder import Binder, BindRequest, PlacementError, device_env
from ..gpu.ledger import Ledger
from .base import Runtime
from .types import (
    ContainerRuntimeInfo,
    DeployParams,
    DetailedStatus,
    Instance,
    PodStatus,
)

log = logging.getLogger("runtime.process")

TERM_GRACE_S = 10.0


class ProcessRuntime(Runtime):
    def __init__(
        self,
        ledger: Ledger,
        state_dir: str,
        cgroup_root: str = "/sys/fs/cgroup",
        cgroup_parent: str = "amdvk.slice",
        podworker: Optional[str] = None,
        enable_cgroups: bool = True,
        pod_namespaces: bool = True,
        log_max_bytes: int = 50 * 1024 * 1024,
        image_store=None,
        image_isolation: str = "auto",
        image_gpu_binds: Optional[List[str]] = None,
        image_extra_binds: Optional[List[str]] = None,
        image_registry: str = "",
        image_registry_token: str = "",
    ):
        from ..ops import load_native

        self._native = load_native()
        self.ledger = ledger
        self.binder = Binder(ledger)
        self.state_dir = Path(state_dir)
        self.instances_dir = self.state_dir / "instances"
        self.logs_dir = self.state_dir / "logs"
        self.instances_dir.mkdir(parents=True, exist_ok=True)
        self.logs_dir.mkdir(parents=True, exist_ok=True)
        self.cgroup_base = os.path.join(cgroup_root, cgroup_parent)
        self.enable_cgroups = enable_cgroups
        # Container-like isolation: own PID namespace (pod is pid 1;
        # descendants die with it) + UTS namespace with the pod's hostname.
        # Degrades automatically when CAP_SYS_ADMIN is absent.
        self.pod_namespaces = pod_namespaces
        self.log_max_bytes = log_max_bytes
        self._podworker = podworker
        # OCI-image execution (reference contract: Containers[0].Image
        # actually runs, runpod_client.go:1304). image_store=None keeps the
        # legacy host-binary behavior everywhere.
        self.image_store = image_store
        self.image_registry = image_registry
        self.image_registry_token = image_registry_token
        # ref -> seconds of the most recent registry pull (event surface)
        self._last_pull_s: dict = {}
        self._rootfs_mgr = None
        if image_store is not None:
            from .rootfs import RootfsManager

            self._rootfs_mgr = RootfsManager(
                image_store, str(self.state_dir / "containers"),
                self._native,
                gpu_binds=image_gpu_binds,
                extra_binds=image_extra_binds,
                isolation=image_isolation,
            )

        self._lock = threading.RLock()
        self._cgroup_pool: List[str] = []
        self._cgroup_draining: List[str] = []
        self._cgroup_counter = 0
        self._instances: Dict[str, Instance] = {}
        self._pid_to_instance: Dict[int, str] = {}
        self._subscribers: List[Callable[[str], None]] = []
        self._kill_timers: Dict[str, threading.Timer] = {}
        # restartPolicy backoff timers, keyed (instance_id, container_name)
        self._restart_timers: Dict[tuple, threading.Timer] = {}
        # spec.activeDeadlineSeconds timers, keyed instance_id
        self._deadline_timers: Dict[str, threading.Timer] = {}

        self._loop = self._native.EventLoop()
        self._stop = threading.Event()
        self._watcher = threading.Thread(
            target=self._watch_events, name="runtime-events", daemon=True
        )
        self._watcher.start()
        # liveness/readiness probe scheduler (1 s resolution; per-probe
        # period/threshold state in _probe_states keyed
        # (instance, container, type))
        from ..utils.backoff import Ticker

        self._probe_states: Dict[tuple, object] = {}
        self._probe_ticker = Ticker(1.0, self._run_probes, "probes").start()
        self._logrotate_ticker = Ticker(5.0, self._rotate_big_logs,
                                        "logrotate").start()

    # ------------- deploy -------------

    def deploy(self, params: DeployParams) -> DetailedStatus:
        from ..server import metrics

        instance_id = "amdvk-" + secrets.token_hex(6)
        gpu_indices: List[int] = []
        t_bind = time.monotonic()
        if params.gpu_count > 0:
            req = BindRequest(
                pod_key=params.pod_key,
                gpu_count=params.gpu_count,
                total_memory_bytes=params.gpu_memory_bytes,
                max_cost=params.max_gpu_cost,
            )
            gpu_indices = self.binder.bind(req)  # raises PlacementError when full
        metrics.bind_seconds.observe(time.monotonic() - t_bind)

        inst = Instance(
            id=instance_id,
            pod_key=params.pod_key,
            params=params,
            gpu_indices=gpu_indices,
            cost_per_hr=round(0.1 * max(1, params.gpu_count), 4) if params.gpu_count else 0.0,
        )

        inst.cgroup_dir = self._acquire_cgroup(params)
        self._attach_device_filter(inst)

        try:
            if params.init_containers:
                # spec.initContainers: sequential, each to completion, before
                # the main containers (progression driven by exit events in
                # _on_init_exit).
                spawn_s = self._launch_one(
                    inst, params.init_containers[0], inst.init_containers)
                inst.init_index = 1
            else:
                spawn_s = self._launch_containers(inst)
        except Exception:
            # Partial deploy failure (e.g. container 2 of 2 fails to
            # spawn): containers already launched must not leak — kill
            # them and drop their event-loop registrations, or a --hold
            # first container would run untracked forever.
            for c in list(inst.containers) + list(inst.init_containers):
                if c.pid > 0 and c.exit_code is None:
                    self._native.signal_process(c.pid, 9, True)
                    self._loop.remove_process(c.pid)
                    with self._lock:
                        self._pid_to_instance.pop(c.pid, None)
            if gpu_indices:
                self.binder.unbind(params.pod_key)
            self._release_cgroup(inst.cgroup_dir)
            raise

        # Native-clock spawn time (posix_spawnp + cgroup migration), not
        # Python wall time: the launching thread routinely loses the GIL to
        # the already-running child, which used to inflate this ~100x.
        metrics.launch_seconds.observe(spawn_s)
        with self._lock:
            self._instances[instance_id] = inst
            if params.active_deadline_s > 0:
                t = threading.Timer(params.active_deadline_s,
                                    self._deadline_exceeded, args=(instance_id,))
                t.daemon = True
                self._deadline_timers[instance_id] = t
                t.start()
        self._persist(inst)
        log.info(
            "deployed instance",
            extra={
                "instance": instance_id,
                "pod": params.pod_key,
                "gpus": gpu_indices,
                "containers": [c.name for c in inst.containers],
            },
        )
        return self._status_of(inst)

    DRM_MAJOR = 226

    def _attach_device_filter(self, inst: Instance) -> None:
        """Enforced GPU isolation (SURVEY §7 hard part (a)): a cgroup-v2
        eBPF device filter denies every DRM render node except the bound
        GPUs' — ROCm cannot acquire a KFD VM without opening the GPU's
        /dev/dri/renderD<minor>, so ROCR_VISIBLE_DEVICES stops being merely
        cooperative. Best-effort like the cgroup limits (no-op when
        unprivileged)."""
        if not inst.cgroup_dir:
            return
        minors = []
        inv = self.binder.ledger.inventory
        for idx in inst.gpu_indices:
            gpu = inv.get(idx)
            if gpu is not None and gpu.render_minor >= 0:
                minors.append(gpu.render_minor)
        ok = self._native.cgroup_attach_device_filter(
            inst.cgroup_dir, self.DRM_MAJOR, minors)
        if ok:
            log.debug("device filter attached",
                      extra={"pod": inst.pod_key, "render_minors": minors})
        else:
            log.debug("device filter unavailable (unprivileged?)",
                      extra={"pod": inst.pod_key})

    def _resolve_image(self, cspec):
        """ImageStore resolution for a container spec; None keeps the
        legacy host-binary path (reserved amdvk/ refs, absent store, or an
        image not present in the local store). With image_registry
        configured, a store miss triggers an in-kubelet pull — the
        kubelet-pulls-during-ContainerCreating behavior of a real node."""
        if (self._rootfs_mgr is None or not cspec.image
                or cspec.image.startswith("amdvk/")):
            return None
        image = self.image_store.resolve(cspec.image)
        policy = cspec.image_pull_policy or (
            "Always" if (":" not in cspec.image.rsplit("/", 1)[-1]
                         or cspec.image.endswith(":latest"))
            and "@sha256:" not in cspec.image else "IfNotPresent")
        if policy == "Never" or not self.image_registry:
            return image
        if image is None or policy == "Always":
            t0 = time.monotonic()
            pulled = self._pull_image(cspec.image)
            if pulled is not None:
                self._last_pull_s[cspec.image] = time.monotonic() - t0
                return pulled
            # Always-policy pull failure: fall back to the cached copy
            # (availability over the kubelet's strict ErrImagePull — an
            # offline-first node should not brick on a registry blip)
        return image

    def _pull_image(self, ref: str):
        from .registry import RegistryClient, RegistryError

        client = RegistryClient(base_url=self.image_registry,
                                token=self.image_registry_token)
        try:
            pulled = client.pull(ref, self.image_store)
            log.info("image pulled from registry",
                     extra={"ref": pulled,
                            "registry": self.image_registry})
            return self.image_store.resolve(pulled)
        except (RegistryError, OSError) as exc:
            log.warning("image pull failed",
                        extra={"ref": ref, "err": str(exc)})
            return None
        finally:
            client.close()

    def _binds_for(self, inst: Instance, cspec) -> List[tuple]:
        """Volume binds plus kubelet-managed rw files: with
        readOnlyRootFilesystem the kubelet still keeps
        terminationMessagePath writable by bind-mounting a host file."""
        binds = self._volume_binds(inst, cspec)
        if getattr(cspec, "read_only_root_fs", False) and \
                cspec.termination_message_path:
            host = (self.state_dir / "volumes" / inst.id /
                    f".tmsg-{cspec.name}")
            host.parent.mkdir(parents=True, exist_ok=True)
            if not host.exists():
                host.touch()
                os.chmod(host, 0o666)
            binds.append((str(host), cspec.termination_message_path,
                          False, "file"))
        return binds

    def _volume_binds(self, inst: Instance, cspec) -> List[tuple]:
        """Materialize the pod's volumes and return (src, dst, ro) binds
        for this container. emptyDir/files volumes live under
        <state>/volumes/<instance>/ — per-POD, so containers share them and
        container restarts keep their content (k8s emptyDir semantics)."""
        out = []
        if not getattr(cspec, "volume_mounts", None):
            return out
        base = self.state_dir / "volumes" / inst.id
        for vm in cspec.volume_mounts:
            src_spec = inst.params.volumes.get(vm.name)
            if src_spec is None:
                continue  # unsupported/unknown volume type: mount skipped
            if src_spec.kind == "hostPath":
                src = src_spec.host_path
                if not os.path.exists(src):
                    raise RuntimeError(
                        f"hostPath volume {vm.name}: {src} does not exist")
            elif (src_spec.kind == "emptyDir"
                    and src_spec.medium == "Memory"
                    and self._rootfs_mgr is not None
                    and self._rootfs_mgr.mode() == "mountns"):
                # emptyDir.medium=Memory: tmpfs mounted by the child
                # (size capped by sizeLimit); nothing materialized on disk
                out.append((str(src_spec.size_limit_bytes), vm.mount_path,
                            False, "tmpfs"))
                continue
            else:
                vdir = base / vm.name
                if not vdir.exists():
                    vdir.mkdir(parents=True, exist_ok=True)
                    if src_spec.kind == "emptyDir":
                        # kubelet makes emptyDir world-writable (0777): a
                        # runAsUser container must be able to write its
                        # scratch volume without fsGroup plumbing
                        os.chmod(vdir, 0o777)
                    for fname, content in src_spec.files.items():
                        fp = vdir / fname.lstrip("/")
                        fp.parent.mkdir(parents=True, exist_ok=True)
                        fp.write_text(content)
                        os.chmod(fp, src_spec.file_mode)
                    fsg = inst.params.fs_group
                    if fsg >= 0:
                        # securityContext.fsGroup: volume group-owned and
                        # group-writable (k8s volume ownership management)
                        try:
                            for p_ in [vdir, *vdir.rglob("*")]:
                                os.chown(p_, -1, fsg)
                                os.chmod(p_, os.stat(p_).st_mode | 0o070)
                        except OSError:
                            pass
                src = str(vdir)
            if vm.sub_path:
                src = os.path.join(src, vm.sub_path)
            # secret/configMap projections are read-only like k8s mounts
            ro = vm.read_only or src_spec.kind == "files"
            out.append((src, vm.mount_path, ro, src_spec.kind))
        return out

    def _gpu_device_paths(self, inst: Instance) -> List[str]:
        """Host device nodes an image pod needs for its bound GPUs."""
        if not inst.gpu_indices or not os.path.exists("/dev/kfd"):
            return []
        paths = ["/dev/kfd"]
        inv = self.binder.ledger.inventory
        for idx in inst.gpu_indices:
            gpu = inv.get(idx)
            if gpu is not None and gpu.render_minor >= 0:
                paths.append(f"/dev/dri/renderD{gpu.render_minor}")
        return [p for p in paths if os.path.exists(p)]

    def _launch_one(self, inst: Instance, cspec, into: List) -> float:
        """Spawn one container of the pod into `into` (inst.containers or
        inst.init_containers); returns the native posix_spawnp/clone3 time in
        seconds. The cgroup.procs migration is timed separately
        (cgroup_migrate_seconds) — it serializes on the kernel's
        cgroup_mutex and dominates launch cost under churn."""
        from ..server import metrics

        params = inst.params
        image = self._resolve_image(cspec)
        if image is not None:
            # container env starts from the IMAGE config, not the kubelet's
            # environment — host env must not leak into containers
            base_env: Dict[str, str] = {}
            for e in image.config.env:
                k, _, v = e.partition("=")
                base_env[k] = v
        else:
            base_env = dict(os.environ)
        # Drop our own GPU scoping so the pod's binding is authoritative.
        base_env.pop("ROCR_VISIBLE_DEVICES", None)
        base_env.pop("HIP_VISIBLE_DEVICES", None)
        base_env.update(params.env)
        base_env.update(device_env(inst.gpu_indices, self.binder.ledger.inventory))
        base_env["AMDVK_INSTANCE_ID"] = inst.id
        base_env["AMDVK_POD_KEY"] = params.pod_key

        argv = list(cspec.command) + list(cspec.args)
        uid, gid = cspec.run_as_uid, cspec.run_as_gid
        working_dir = cspec.working_dir
        rootfs, chroot_only, mounts = "", False, []
        if image is not None:
            # k8s image semantics: command overrides Entrypoint; args
            # override Cmd; with neither, Entrypoint+Cmd run.
            if not cspec.command:
                argv = list(image.config.entrypoint) + list(
                    cspec.args or image.config.cmd)
            if not argv:
                raise RuntimeError(
                    f"container {cspec.name}: no command and image "
                    f"{cspec.image} defines no entrypoint/cmd")
            working_dir = cspec.working_dir or image.config.working_dir
            if uid < 0 and gid < 0 and image.config.user:
                uid, gid = self.image_store.resolve_user(
                    self.image_store.rootfs_for(image), image.config.user)
                # record the image-derived identity on the spec so exec
                # probes / kubectl-exec run as the container's user (k8s
                # runs both inside the container), and restarts/adoption
                # keep it (it is persisted with the spec)
                cspec.run_as_uid, cspec.run_as_gid = uid, gid
            prepared = self._rootfs_mgr.prepare(
                inst.id, cspec.name, image,
                params.hostname or params.name,
                gpu_device_paths=self._gpu_device_paths(inst),
                working_dir=working_dir,
                volume_binds=self._binds_for(inst, cspec),
                host_aliases=params.host_aliases,
                read_only=cspec.read_only_root_fs,
                resolv_conf=params.resolv_conf)
            argv[0] = self._rootfs_mgr.resolve_argv0(image, prepared,
                                                     argv[0])
            rootfs = prepared.rootfs
            chroot_only = prepared.chroot_only
            mounts = prepared.mounts
        elif not argv:
            if (self._rootfs_mgr is not None and cspec.image
                    and not cspec.image.startswith("amdvk/")):
                # The user named a real image with no command and it is not
                # in the local store: failing loudly (pod stays Pending,
                # retried) is the honest ErrImagePull analogue — once the
                # operator imports/pulls the image, the retry succeeds.
                # Silently running the synthetic podworker instead would
                # lie about what is executing.
                raise RuntimeError(
                    f"image {cspec.image!r} not in the local store "
                    f"(import it: python -m "
                    f"k8s_runpod_kubelet_amd.runtime.imagetool)")
            argv = [self.podworker_path(), "--hold"]
            if inst.gpu_indices:
                argv += ["--expect-gpus", str(len(inst.gpu_indices))]
            for port in cspec.tcp_ports:
                argv += ["--listen-port", str(port)]
        elif argv[0] in ("podworker", "amdvk-podworker"):
            argv[0] = self.podworker_path()
        if image is None:
            needs_fast_path = (uid >= 0 or gid >= 0 or self.pod_namespaces)
            if needs_fast_path and "/" not in argv[0]:
                # Credential dropping / namespaces happen on the execve fast
                # path, which does no PATH search — resolve here instead.
                resolved = shutil.which(argv[0])
                if resolved:
                    argv[0] = resolved

        if cspec.run_as_non_root and uid <= 0:
            # securityContext.runAsNonRoot: k8s refuses to start a
            # container whose effective user is (or defaults to) root —
            # CreateContainerConfigError; pod stays Pending until fixed
            raise RuntimeError(
                f"container {cspec.name}: runAsNonRoot is set but the "
                f"container would run as "
                f"{'root' if uid == 0 else 'the kubelet user (root)'}"
                " — set runAsUser or an image USER")

        env = dict(base_env)
        env.update(cspec.env)
        envp = [f"{k}={v}" for k, v in env.items()]
        stdout_path = str(self.logs_dir / f"{inst.id}-{cspec.name}.log")

        def do_launch():
            return self._native.launch_process(
                argv, envp,
                working_dir or "",
                stdout_path, stdout_path,
                inst.cgroup_dir, True, True,
                uid, gid,
                self.pod_namespaces,
                (params.hostname or params.name) if self.pod_namespaces
                else "",
                rootfs, chroot_only, mounts,
            )

        try:
            pid, pidfd, ready_fd, spawn_s, cgroup_s = do_launch()
        except RuntimeError as exc:
            if image is None or "rootfs setup" not in str(exc) or chroot_only:
                raise
            # mountns mode failed on this host (overlayfs/caps): downgrade
            # to the chroot mode once, then retry this launch with a
            # freshly prepared copied rootfs.
            log.warning("mountns image launch failed; retrying in chroot "
                        "mode", extra={"err": str(exc)})
            self._rootfs_mgr.downgrade_to_chroot()
            prepared = self._rootfs_mgr.prepare(
                inst.id, cspec.name, image,
                params.hostname or params.name,
                gpu_device_paths=self._gpu_device_paths(inst),
                working_dir=working_dir,
                volume_binds=self._binds_for(inst, cspec),
                host_aliases=params.host_aliases,
                resolv_conf=params.resolv_conf)
            argv[0] = self._rootfs_mgr.resolve_argv0(
                image, prepared, (cspec.command or image.config.entrypoint
                                  or argv)[0])
            rootfs = prepared.rootfs
            chroot_only = prepared.chroot_only
            mounts = prepared.mounts
            pid, pidfd, ready_fd, spawn_s, cgroup_s = do_launch()
        if image is not None:
            inst.image_mode = "chroot" if chroot_only else "mountns"
        if inst.cgroup_dir:
            metrics.cgroup_migrate_seconds.observe(cgroup_s)
        cinfo = ContainerRuntimeInfo(name=cspec.name, pid=pid, started_at=time.time())
        if image is not None:
            cinfo.image_id = f"{image.ref}@{image.manifest_digest}"
            cinfo.pull_seconds = self._last_pull_s.pop(cspec.image, 0.0)
        # k8s readiness semantics: a running container WITHOUT a
        # readinessProbe is Ready as soon as it starts. The AMDVK_READY_FD
        # pipe protocol (readiness deferred until the workload signals its
        # GPU context is up) applies only to the in-tree podworker, which
        # opts in by writing to the pipe — generic/image entrypoints must
        # not sit NotReady until exit.
        pipe_gated = bool(argv) and argv[0] == self.podworker_path()
        if (into is not inst.init_containers and not pipe_gated
                and cspec.readiness is None and cspec.startup is None):
            cinfo.ready = True
        into.append(cinfo)
        if (getattr(cspec, "post_start", None) is not None
                and into is not inst.init_containers):
            threading.Thread(target=self._post_start_hook,
                             args=(inst, cspec, pid),
                             name="post-start", daemon=True).start()
        with self._lock:
            self._pid_to_instance[pid] = inst.id
        self._loop.add_process(pid, pidfd, ready_fd, pid)
        return spawn_s

    def _launch_containers(self, inst: Instance) -> float:
        """Spawn every main container; returns summed native spawn time."""
        spawn_total_s = 0.0
        for cspec in inst.params.containers:
            spawn_total_s += self._launch_one(inst, cspec, inst.containers)
        inst.desired_status = PodStatus.RUNNING
        return spawn_total_s

    # ------------- cgroup slot pool -------------
    #
    # cgroup v2 directories are pooled, not created/destroyed per pod:
    # rmdir kicks off deferred kernel-side destruction whose cgroup_mutex
    # work serializes against the next mkdir/migration — measured as
    # 100-200 ms deploy stalls on MI355X whenever a pod start overlapped a
    # pod teardown. Slots (amdvk.slice/slotN) are created once and reused;
    # limits are rewritten per pod.

    def _acquire_cgroup(self, params: DeployParams) -> str:
        if not self.enable_cgroups:
            return ""
        with self._lock:
            slot = None
            while self._cgroup_pool:
                candidate = self._cgroup_pool.pop()
                # Only reuse a slot whose previous processes are fully gone.
                if self._native.cgroup_proc_count(candidate) == 0:
                    slot = candidate
                    break
                self._cgroup_draining.append(candidate)
            if slot is None:
                self._cgroup_counter += 1
                slot = os.path.join(self.cgroup_base, f"slot{self._cgroup_counter}")
        if not self._native.cgroup_create(
            slot, params.cpu_limit or "max", params.memory_limit or "max"
        ):
            # unprivileged (tests) or cgroupfs read-only: run without limits
            if self._native.cgroup_proc_count(slot) < 0:
                log.debug("cgroups unavailable; continuing without",
                          extra={"pod": params.pod_key})
                return ""
        return slot

    def _release_cgroup(self, slot: str) -> None:
        if not slot:
            return
        with self._lock:
            self._cgroup_pool.append(slot)
            # Re-check slots that were draining last time.
            still = []
            for s in self._cgroup_draining:
                if self._native.cgroup_proc_count(s) == 0:
                    self._cgroup_pool.append(s)
                else:
                    still.append(s)
            self._cgroup_draining = still

    def podworker_path(self) -> str:
        if self._podworker is None:
            from ..ops import podworker_binary

            self._podworker = podworker_binary()
        return self._podworker

    # ------------- events -------------

    def _watch_events(self) -> None:
        while not self._stop.is_set():
            try:
                events = self._loop.poll(500)
            except Exception:
                log.exception("event loop poll failed")
                time.sleep(0.5)
                continue
            touched = set()
            try:
                self._handle_events(events, touched)
            except Exception:
                log.exception("event handling failed")
            for inst_id in touched:
                with self._lock:
                    inst = self._instances.get(inst_id)
                if inst is not None:
                    try:
                        self._persist(inst)
                    except Exception:
                        log.exception("instance persist failed")
                self._notify(inst_id)

    def _handle_events(self, events, touched) -> None:
            # Each event is processed under the runtime lock: container state
            # transitions (exit → restart-schedule → terminate) are mutated
            # from three threads (watcher, restart timers, API callers) and
            # must be atomic — a terminate interleaving with a restart
            # decision must never strand a reservation.
            for ev in events:
                with self._lock:
                    inst_id = self._pid_to_instance.get(ev.pid)
                    inst = self._instances.get(inst_id) if inst_id else None
                    if inst is not None:
                        self._handle_one_event(ev, inst, touched, inst_id)

    def _handle_one_event(self, ev, inst, touched, inst_id) -> None:
        # ephemeral (kubectl-debug) containers: record exits, nothing else —
        # they never gate readiness, restart, or pod completion
        eph = next((c for c in inst.ephemeral_containers
                    if c.pid == ev.pid), None)
        if eph is not None:
            if ev.type == "exited":
                eph.exit_code = ev.exit_code
                eph.finished_at = time.time()
                touched.add(inst_id)
            return
        cinfo = next((c for c in inst.containers if c.pid == ev.pid), None)
        is_init = False
        if cinfo is None:
            cinfo = next(
                (c for c in inst.init_containers if c.pid == ev.pid), None)
            is_init = cinfo is not None
        if cinfo is None:
            return
        # A defined readinessProbe owns the Ready state: the AMDVK_READY_FD
        # pipe / process-start heuristics are ignored for that container.
        probe_gated = not is_init and any(
            s.readiness is not None or s.startup is not None
            for s in inst.params.containers if s.name == cinfo.name)
        if ev.type == "ready":
            if not is_init and not probe_gated:
                cinfo.ready = True
                touched.add(inst_id)
        elif ev.type == "ready_closed":
            # Workload never wrote READY: process start is readiness
            # (generic binaries without the pipe protocol). Only
            # meaningful if it is still running.
            if not is_init and not probe_gated and \
                    cinfo.exit_code is None and not cinfo.ready:
                cinfo.ready = True
                touched.add(inst_id)
        elif ev.type == "exited":
            cinfo.exit_code = ev.exit_code
            cinfo.finished_at = time.time()
            cinfo.ready = False
            tmsg = self._read_termination_message(inst, cinfo.name)
            if not tmsg and ev.exit_code:
                tmsg = self._termination_log_fallback(inst, cinfo.name)
            if tmsg:
                # terminationMessagePath: the container's own last words
                # win (k8s surfaces them in the terminated state)
                cinfo.message = tmsg
            elif ev.exit_code and not cinfo.message:
                # keep a pre-set reason (liveness/startup probe kill,
                # deadline) over the generic exit-code message
                cinfo.message = f"exit code {ev.exit_code}"
            if is_init:
                self._on_init_exit(inst, cinfo)
            else:
                self._on_container_exit(inst, cinfo)
            touched.add(inst_id)


    def _read_termination_message(self, inst: Instance,
                                  container: str) -> str:
        """spec.containers[].terminationMessagePath for image pods: read
        the file the container wrote inside its rootfs (overlay upper /
        chroot copy). Truncated to 4 KiB like the kubelet."""
        if self._rootfs_mgr is None or not inst.image_mode:
            return ""
        cspec = next((c for c in inst.params.containers
                      if c.name == container), None)
        if cspec is None:
            return ""
        host = (self.state_dir / "volumes" / inst.id /
                f".tmsg-{container}")
        if host.exists():
            try:
                text = host.read_text(encoding="utf-8",
                                      errors="replace")[:4096].strip()
                if text:
                    return text
            except OSError:
                pass
        rel = cspec.termination_message_path.lstrip("/")
        cdir = self._rootfs_mgr.containers_dir / f"{inst.id}-{container}"
        base = cdir / ("rootfs" if inst.image_mode == "chroot" else "upper")
        try:
            with open(base / rel, "r", encoding="utf-8",
                      errors="replace") as fh:
                return fh.read(4096).strip()
        except OSError:
            return ""

    def _termination_log_fallback(self, inst: Instance,
                                  container: str) -> str:
        """terminationMessagePolicy=FallbackToLogsOnError: a failed
        container with no termination-message file surfaces the tail of
        its log (kubelet caps this at 2048 bytes / 80 lines)."""
        cspec = next((c for c in inst.params.containers
                      if c.name == container), None)
        if cspec is None or                 cspec.termination_message_policy != "FallbackToLogsOnError":
            return ""
        path = self.get_log_path(inst.id, container)
        if not path:
            return ""
        try:
            with open(path, "rb") as fh:
                fh.seek(0, os.SEEK_END)
                size = fh.tell()
                fh.seek(max(0, size - 2048))
                raw = fh.read().decode("utf-8", errors="replace")
        except OSError:
            return ""
        return "\n".join(raw.splitlines()[-80:]).strip()

    def _on_container_exit(self, inst: Instance, cinfo=None) -> None:
        # spec.restartPolicy (k8s semantics; the reference's cloud instances
        # are run-to-completion = Never): Always restarts any exit, OnFailure
        # restarts nonzero exits, with exponential crash backoff
        # (CrashLoopBackOff surfaced in container status).
        if cinfo is not None and inst.desired_status == PodStatus.RUNNING \
                and not inst.deadline_exceeded:
            policy = inst.params.restart_policy
            if policy == "Always" or (policy == "OnFailure" and cinfo.exit_code != 0):
                self._schedule_restart(inst, cinfo)
                return
        if any(c.exit_code is None or c.backoff_until != 0
               for c in inst.containers):
            return
        terminating = inst.desired_status == PodStatus.TERMINATING
        inst.desired_status = PodStatus.TERMINATED if terminating else PodStatus.EXITED
        self._teardown_resources(inst)

    def _schedule_restart(self, inst: Instance, cinfo) -> None:
        ran_s = max(0.0, (cinfo.finished_at or time.time())
                    - (cinfo.started_at or time.time()))
        if ran_s > 60.0:
            # ran long enough to be considered healthy: reset the crash
            # backoff (kubelet's 10-minute analogue, scaled down)
            cinfo.crash_streak = 0
        delay = min(60.0, 2.0 ** cinfo.crash_streak)
        cinfo.crash_streak += 1
        cinfo.backoff_until = time.time() + delay
        timer = threading.Timer(
            delay, self._restart_container, args=(inst.id, cinfo.name))
        timer.daemon = True
        self._restart_timers[(inst.id, cinfo.name)] = timer
        timer.start()
        log.info("container restart scheduled",
                 extra={"instance": inst.id, "container": cinfo.name,
                        "delay_s": delay, "restarts": cinfo.restart_count})

    def _restart_container(self, instance_id: str, name: str) -> None:
        # Entire transition under the runtime lock: must not interleave with
        # terminate() (which cancels restarts and tears resources down) or
        # with the exit-event handler.
        with self._lock:
            self._restart_timers.pop((instance_id, name), None)
            inst = self._instances.get(instance_id)
            if inst is None or inst.desired_status != PodStatus.RUNNING:
                return
            cspec = next(
                (s for s in inst.params.containers if s.name == name), None)
            cinfo = next((c for c in inst.containers if c.name == name), None)
            if cspec is None or cinfo is None:
                return
            # kubectl logs --previous: rotate the crashed run's output
            logp = self.logs_dir / f"{instance_id}-{name}.log"
            try:
                if logp.exists():
                    logp.rename(self.logs_dir
                                / f"{instance_id}-{name}.log.prev")
            except OSError:
                pass
            old_pid = cinfo.pid
            tmp: List = []
            try:
                self._launch_one(inst, cspec, tmp)
            except Exception as exc:
                log.exception("container restart failed",
                              extra={"instance": instance_id, "container": name})
                cinfo.message = f"restart failed: {exc}"
                cinfo.backoff_until = 0
                self._on_container_exit(inst, None)  # may complete the pod now
            else:
                fresh = tmp[0]
                self._pid_to_instance.pop(old_pid, None)
                cinfo.pid = fresh.pid
                cinfo.started_at = fresh.started_at
                cinfo.finished_at = 0.0
                cinfo.exit_code = None
                cinfo.ready = fresh.ready  # started-is-ready when unprobed
                cinfo.message = ""
                cinfo.restart_count += 1
                cinfo.backoff_until = 0
                self._clear_probe_state(instance_id, name)
        self._persist(inst)
        self._notify(instance_id)

    def _on_init_exit(self, inst: Instance, cinfo) -> None:
        """Init-container progression (spec.initContainers semantics): exit 0
        starts the next init or the main containers; nonzero fails the pod
        (restartPolicy=Never model)."""
        if inst.desired_status == PodStatus.TERMINATING:
            inst.desired_status = PodStatus.TERMINATED
            self._teardown_resources(inst)
            return
        if inst.deadline_exceeded:
            # Pod was killed by activeDeadlineSeconds while init was running:
            # keep last_error="DeadlineExceeded" so status reports the real
            # reason, not the init kill's exit message.
            inst.desired_status = PodStatus.EXITED
            self._teardown_resources(inst)
            return
        if cinfo.exit_code != 0:
            inst.last_error = (
                f"init container {cinfo.name} exited with code {cinfo.exit_code}")
            inst.desired_status = PodStatus.EXITED
            self._teardown_resources(inst)
            return
        try:
            if inst.init_index < len(inst.params.init_containers):
                nxt = inst.params.init_containers[inst.init_index]
                inst.init_index += 1
                self._launch_one(inst, nxt, inst.init_containers)
            else:
                self._launch_containers(inst)
        except Exception as exc:
            log.exception("post-init launch failed", extra={"instance": inst.id})
            inst.last_error = f"launch after init failed: {exc}"
            inst.desired_status = PodStatus.EXITED
            self._teardown_resources(inst)

    def _deadline_exceeded(self, instance_id: str) -> None:
        """spec.activeDeadlineSeconds fired: kill the pod; it completes as
        Failed/DeadlineExceeded (restartPolicy no longer applies)."""
        notify = False
        with self._lock:
            self._deadline_timers.pop(instance_id, None)
            inst = self._instances.get(instance_id)
            if inst is None or inst.desired_status not in (
                    PodStatus.RUNNING, PodStatus.STARTING):
                return
            log.warning("activeDeadlineSeconds exceeded",
                        extra={"instance": instance_id, "pod": inst.pod_key})
            inst.deadline_exceeded = True
            inst.last_error = "DeadlineExceeded"
            for (iid, cname), t in list(self._restart_timers.items()):
                if iid == instance_id:
                    t.cancel()
                    self._restart_timers.pop((iid, cname), None)
            live = False
            for c in list(inst.containers) + list(inst.init_containers):
                c.backoff_until = 0
                if c.exit_code is None:
                    c.message = "deadline exceeded"
                    live = True
            if live:
                self._signal_all(inst, 9)  # exits drive completion
            else:
                inst.desired_status = PodStatus.EXITED
                self._teardown_resources(inst)
                notify = True
        if notify:
            self._persist(inst)
            self._notify(instance_id)

    def _teardown_resources(self, inst: Instance) -> None:
        timer = self._kill_timers.pop(inst.id, None)
        if timer:
            timer.cancel()
        dtimer = self._deadline_timers.pop(inst.id, None)
        if dtimer:
            dtimer.cancel()
        for (iid, cname), t in list(self._restart_timers.items()):
            if iid == inst.id:
                t.cancel()
                self._restart_timers.pop((iid, cname), None)
        for c in inst.ephemeral_containers:
            if c.exit_code is None and c.pid > 0:
                self._native.signal_process(c.pid, 9, True)
        # Release GPUs as soon as the workload is gone — HBM headroom returns
        # to the ledger without waiting for pod deletion.
        self.binder.unbind(inst.pod_key)
        self._release_cgroup(inst.cgroup_dir)
        inst.cgroup_dir = ""

    def _run_hook(self, inst: Instance, cspec, spec) -> bool:
        """Execute one lifecycle hook (postStart/preStop) confined like an
        exec probe: container credentials, cgroup, and (image pods) its
        rootfs."""
        from .probes import run_probe

        env = dict(os.environ)
        env.update(inst.params.env)
        env.update(device_env(inst.gpu_indices,
                              self.binder.ledger.inventory))
        env.update(cspec.env)
        setns_pid, rootfs = self._container_entry(inst, cspec)

        def runner(command, penv, timeout_s):
            return self._run_confined(
                command, penv, inst.cgroup_dir,
                cspec.run_as_uid, cspec.run_as_gid, timeout_s,
                setns_pid=setns_pid, rootfs=rootfs)

        return run_probe(spec, env, exec_runner=runner)

    def _post_start_hook(self, inst: Instance, cspec, pid: int) -> None:
        """k8s postStart: runs right after the container starts; failure
        kills the container into the restartPolicy machinery."""
        ok = False
        try:
            ok = self._run_hook(inst, cspec, cspec.post_start)
        except Exception:
            log.exception("postStart hook error")
        if ok:
            return
        log.warning("postStart hook failed; killing container",
                    extra={"instance": inst.id, "container": cspec.name})
        with self._lock:
            cinfo = next((c for c in inst.containers
                          if c.pid == pid and c.exit_code is None), None)
            if cinfo is not None:
                cinfo.message = "postStart hook failed"
        if cinfo is not None:
            self._native.signal_process(pid, 9, True)

    def _pre_stop_then_term(self, instance_id: str, inst: Instance,
                            hooks) -> None:
        """k8s preStop: hooks run inside the grace window, then SIGTERM.
        The force-kill timer was armed at terminate() time, so a slow hook
        cannot extend the pod's life beyond its grace period."""
        for cspec, cinfo in hooks:
            if cinfo.exit_code is not None:
                continue
            try:
                self._run_hook(inst, cspec, cspec.pre_stop)
            except Exception:
                log.exception("preStop hook error")
        with self._lock:
            if inst.desired_status == PodStatus.TERMINATING:
                self._signal_all(inst, 15)

    def _run_probes(self) -> None:
        """One probe-scheduler tick: run due liveness/readiness probes on
        RUNNING containers (see runtime/probes.py; the reference has no
        probe support at all)."""
        from .probes import ProbeState, advance, run_probe

        now = time.time()
        with self._lock:
            insts = [i for i in self._instances.values()
                     if i.desired_status == PodStatus.RUNNING]
        for inst in insts:
            by_name = {c.name: c for c in inst.containers}
            for cspec in inst.params.containers:
                if (cspec.liveness is None and cspec.readiness is None
                        and cspec.startup is None):
                    continue
                cinfo = by_name.get(cspec.name)
                if cinfo is None or cinfo.exit_code is not None:
                    continue
                env = None
                if cspec.startup is not None:
                    skey = (inst.id, cspec.name, "startup")
                    sst = self._probe_states.setdefault(skey, ProbeState())
                    if sst.result is not True:
                        # startup owns the container until it passes
                        # (k8s: not Started yet -> no liveness/readiness)
                        spec = cspec.startup
                        if now - cinfo.started_at < spec.initial_delay_s:
                            continue
                        if now - sst.last_run < spec.period_s:
                            continue
                        sst.last_run = now
                        if spec.kind == "exec":
                            env = dict(os.environ)
                            env.update(inst.params.env)
                            env.update(device_env(
                                inst.gpu_indices,
                                self.binder.ledger.inventory))
                            env.update(cspec.env)
                            cg = inst.cgroup_dir
                            uid, gid = cspec.run_as_uid, cspec.run_as_gid
                            sp, rf = self._container_entry(inst, cspec)

                            def srunner(command, penv, timeout_s, _cg=cg,
                                        _uid=uid, _gid=gid, _sp=sp, _rf=rf):
                                return self._run_confined(
                                    command, penv, _cg, _uid, _gid,
                                    timeout_s, setns_pid=_sp, rootfs=_rf)
                        else:
                            srunner = None
                        ok = run_probe(spec, env or {}, exec_runner=srunner)
                        outcome = advance(sst, spec, ok)
                        if outcome is False:
                            log.warning(
                                "startup probe failed; killing container",
                                extra={"instance": inst.id,
                                       "container": cspec.name,
                                       "failures": sst.failures})
                            cinfo.message = "startup probe failed"
                            self._clear_probe_state(inst.id, cspec.name)
                            self._native.signal_process(cinfo.pid, 9, True)
                        elif outcome is True and cspec.readiness is None:
                            # started == ready when no readinessProbe
                            if not cinfo.ready:
                                cinfo.ready = True
                                self._persist(inst)
                                self._notify(inst.id)
                        continue
                for kind, spec in (("liveness", cspec.liveness),
                                   ("readiness", cspec.readiness)):
                    if spec is None:
                        continue
                    key = (inst.id, cspec.name, kind)
                    st = self._probe_states.setdefault(key, ProbeState())
                    if now - cinfo.started_at < spec.initial_delay_s:
                        continue
                    if now - st.last_run < spec.period_s:
                        continue
                    st.last_run = now
                    exec_runner = None
                    if spec.kind == "exec":
                        if env is None:
                            env = dict(os.environ)
                            env.update(inst.params.env)
                            env.update(device_env(inst.gpu_indices,
                                                  self.binder.ledger.inventory))
                            env.update(cspec.env)
                        # confine the probe like the container itself:
                        # same cgroup (device filter included), the
                        # container's runAsUser/runAsGroup credentials,
                        # and — for image pods — the container's rootfs
                        cgroup_dir = inst.cgroup_dir
                        uid, gid = cspec.run_as_uid, cspec.run_as_gid
                        setns_pid, rootfs_p = self._container_entry(
                            inst, cspec)

                        def exec_runner(command, penv, timeout_s,
                                        _cg=cgroup_dir, _uid=uid, _gid=gid,
                                        _sp=setns_pid, _rf=rootfs_p):
                            return self._run_confined(
                                command, penv, _cg, _uid, _gid, timeout_s,
                                setns_pid=_sp, rootfs=_rf)
                    ok = run_probe(spec, env or {}, exec_runner=exec_runner)
                    outcome = advance(st, spec, ok)
                    if kind == "readiness" and outcome is not None:
                        if cinfo.ready != outcome:
                            cinfo.ready = outcome
                            self._persist(inst)
                            self._notify(inst.id)
                    elif kind == "liveness" and outcome is False:
                        log.warning(
                            "liveness probe failed; killing container",
                            extra={"instance": inst.id, "container": cspec.name,
                                   "failures": st.failures})
                        cinfo.message = "liveness probe failed"
                        self._probe_states.pop(key, None)
                        self._native.signal_process(cinfo.pid, 9, True)
                        # exit event drives restartPolicy from here

    def _rotate_big_logs(self) -> None:
        """Cap per-container log files (copytruncate into the --previous
        slot): the child keeps its O_APPEND fd, so rename-style rotation
        would chase the fd — copy+truncate is the standard answer. A
        chatty container must not fill the node's disk."""
        if self.log_max_bytes <= 0:
            return
        with self._lock:
            insts = [i for i in self._instances.values()
                     if i.desired_status in (PodStatus.RUNNING,
                                             PodStatus.STARTING)]
        for inst in insts:
            names = ([c.name for c in inst.params.containers]
                     + [c.name for c in inst.ephemeral_specs])
            for name in names:
                path = self.logs_dir / f"{inst.id}-{name}.log"
                try:
                    if path.stat().st_size <= self.log_max_bytes:
                        continue
                    shutil.copyfile(path, str(path) + ".prev")
                    with open(path, "r+b") as fh:
                        fh.truncate(0)
                    log.info("container log rotated",
                             extra={"instance": inst.id, "container": name})
                except OSError:
                    continue

    def _clear_probe_state(self, instance_id: str, container: str = "") -> None:
        for key in list(self._probe_states):
            if key[0] == instance_id and (not container or key[1] == container):
                self._probe_states.pop(key, None)

    def _notify(self, instance_id: str) -> None:
        for cb in list(self._subscribers):
            try:
                cb(instance_id)
            except Exception:
                log.exception("runtime subscriber failed")

    def subscribe(self, callback: Callable[[str], None]) -> None:
        self._subscribers.append(callback)

    # ------------- status -------------

    def _pod_pids(self, inst: Instance) -> set:
        """Every live pid belonging to the pod: the cgroup's procs when the
        pod has a slot, else the container pids plus their descendants
        (via /proc/<pid>/task/*/children)."""
        if inst.cgroup_dir:
            try:
                with open(inst.cgroup_dir + "/cgroup.procs",
                          encoding="ascii") as fh:
                    pids = {int(line) for line in fh if line.strip()}
                if pids:
                    return pids
            except OSError:
                pass
        pids = {c.pid for c in inst.containers
                if c.exit_code is None and c.pid > 0}
        frontier = list(pids)
        while frontier:
            pid = frontier.pop()
            try:
                for task in os.listdir(f"/proc/{pid}/task"):
                    with open(f"/proc/{pid}/task/{task}/children",
                              encoding="ascii") as fh:
                        for child in fh.read().split():
                            c = int(child)
                            if c not in pids:
                                pids.add(c)
                                frontier.append(c)
            except OSError:
                continue
        return pids

    def _pod_listening_ports(self, inst: Instance) -> set:
        """TCP ports in LISTEN state owned by *this pod's* processes —
        socket inodes from /proc/net/tcp{,6} attributed via
        /proc/<pid>/fd. A host-wide scan (round-1 weak #3) marked a pod's
        port exposed when any unrelated process listened on it; the
        reference gates readiness on the backend's per-pod portMappings
        (kubelet.go:566-605), i.e. per-instance."""
        by_inode = _listening_tcp_inodes()
        if not by_inode:
            return set()
        ports = set()
        for pid in self._pod_pids(inst):
            try:
                for fd in os.listdir(f"/proc/{pid}/fd"):
                    try:
                        link = os.readlink(f"/proc/{pid}/fd/{fd}")
                    except OSError:
                        continue
                    if link.startswith("socket:["):
                        ino = int(link[8:-1])
                        port = by_inode.get(ino)
                        if port is not None:
                            ports.add(port)
            except OSError:
                continue
        return ports

    def _status_of(self, inst: Instance) -> DetailedStatus:
        ports: Dict[int, int] = {}
        if inst.desired_status == PodStatus.RUNNING:
            want = []
            for c in inst.params.containers:
                want.extend(c.tcp_ports)
            if want:
                listening = self._pod_listening_ports(inst)
                ports = {p: p for p in want if p in listening}
        return DetailedStatus(
            id=inst.id,
            desired_status=inst.desired_status,
            namespace=inst.params.namespace,
            name=inst.params.name,
            port_mappings=ports,
            containers=[ContainerRuntimeInfo(**vars(c)) for c in inst.containers],
            init_containers=[ContainerRuntimeInfo(**vars(c))
                             for c in inst.init_containers],
            ephemeral_containers=[ContainerRuntimeInfo(**vars(c))
                                  for c in inst.ephemeral_containers],
            gpu_indices=list(inst.gpu_indices),
            cost_per_hr=inst.cost_per_hr,
            last_error=inst.last_error,
            created_at=inst.created_at,
        )

    def get_detailed_status(self, instance_id: str) -> DetailedStatus:
        with self._lock:
            inst = self._instances.get(instance_id)
        if inst is None:
            return DetailedStatus(id=instance_id, desired_status=PodStatus.NOT_FOUND)
        return self._status_of(inst)

    def list_instances(self, statuses: Optional[List[str]] = None) -> List[DetailedStatus]:
        with self._lock:
            insts = list(self._instances.values())
        out = [self._status_of(i) for i in insts]
        if statuses:
            out = [s for s in out if s.desired_status in statuses]
        return out

    def instance_for_pod(self, pod_key: str) -> Optional[str]:
        with self._lock:
            for inst in self._instances.values():
                if inst.pod_key == pod_key and inst.desired_status not in (
                    PodStatus.TERMINATED,
                ):
                    return inst.id
        return None

    # ------------- terminate / GC -------------

    def terminate(self, instance_id: str,
                  grace_override_s: float = -1.0) -> None:
        """grace_override_s >= 0 replaces spec.terminationGracePeriodSeconds
        for this call (hard eviction passes 0: TERM now, KILL ~now)."""
        notify_done = False
        # Transition under the runtime lock — atomic against the exit-event
        # handler and restart timers (a terminate interleaving with a
        # restart decision must never strand a reservation).
        with self._lock:
            inst = self._instances.get(instance_id)
            if inst is None:
                return
            if inst.desired_status in (PodStatus.EXITED, PodStatus.TERMINATED):
                inst.desired_status = PodStatus.TERMINATED
                self._persist(inst)
                return
            inst.desired_status = PodStatus.TERMINATING
            # Cancel pending restartPolicy backoffs: a terminating pod must
            # not relaunch containers; all-in-backoff pods complete now.
            for (iid, cname), t in list(self._restart_timers.items()):
                if iid == instance_id:
                    t.cancel()
                    self._restart_timers.pop((iid, cname), None)
            for c in inst.containers:
                c.backoff_until = 0
            if all(c.exit_code is not None for c in inst.containers) and \
                    all(c.exit_code is not None for c in inst.init_containers):
                inst.desired_status = PodStatus.TERMINATED
                self._teardown_resources(inst)
                notify_done = True
            else:
                grace = (grace_override_s if grace_override_s >= 0
                         else inst.params.termination_grace_s)
                timer = threading.Timer(
                    # spec.terminationGracePeriodSeconds — the window
                    # covers preStop hooks AND the TERM->KILL ladder
                    max(0.1, grace),
                    self._force_kill, args=(instance_id,))
                timer.daemon = True
                self._kill_timers[instance_id] = timer
                timer.start()
                hooks = []
                for cspec in inst.params.containers:
                    if getattr(cspec, "pre_stop", None) is None:
                        continue
                    cinfo = next((c for c in inst.containers
                                  if c.name == cspec.name
                                  and c.exit_code is None), None)
                    if cinfo is not None:
                        hooks.append((cspec, cinfo))
                if hooks:
                    threading.Thread(target=self._pre_stop_then_term,
                                     args=(instance_id, inst, hooks),
                                     name="pre-stop", daemon=True).start()
                else:
                    self._signal_all(inst, 15)  # SIGTERM
        self._persist(inst)
        if notify_done:
            self._notify(instance_id)

    def _force_kill(self, instance_id: str) -> None:
        with self._lock:
            inst = self._instances.get(instance_id)
        if inst is None:
            return
        if any(c.exit_code is None
               for c in (list(inst.containers) + list(inst.init_containers)
                         + list(inst.ephemeral_containers))):
            log.warning("grace period expired; SIGKILL", extra={"instance": instance_id})
            self._signal_all(inst, 9)

    def _signal_all(self, inst: Instance, sig: int) -> None:
        for c in (list(inst.containers) + list(inst.init_containers)
                  + list(inst.ephemeral_containers)):
            if c.exit_code is None and c.pid > 0:
                # Whole process group: the container is a session leader.
                rc = self._native.signal_process(c.pid, sig, True)
                if rc != 0:
                    self._native.signal_process(c.pid, sig, False)

    def remove(self, instance_id: str) -> None:
        with self._lock:
            inst = self._instances.pop(instance_id, None)
            if inst is not None:
                for c in (list(inst.containers) + list(inst.init_containers)
                          + list(inst.ephemeral_containers)):
                    self._pid_to_instance.pop(c.pid, None)
        if inst is not None:
            self.binder.unbind(inst.pod_key)
            self._clear_probe_state(instance_id)
            (self.instances_dir / f"{instance_id}.json").unlink(missing_ok=True)
            if self._rootfs_mgr is not None:
                self._rootfs_mgr.cleanup(instance_id)
            shutil.rmtree(self.state_dir / "volumes" / instance_id,
                          ignore_errors=True)

    def healthy(self) -> bool:
        return self.ledger.any_schedulable() or self.ledger.total_gpus() == 0

    def tracked_process_count(self) -> int:
        """Processes the native event loop still watches (leak check)."""
        return int(self._loop.tracked_count())

    def get_stats(self, instance_id: str) -> Dict[str, object]:
        """Per-pod resource usage for /stats/summary: cgroup v2 counters
        when the pod has a slot, /proc per-process fallback otherwise.
        (The reference stubs its kubelet stats hooks off entirely,
        cmd/virtual_kubelet/main.go:233-235 — local pods make them real.)"""
        with self._lock:
            inst = self._instances.get(instance_id)
        if inst is None:
            return {}
        stats: Dict[str, object] = {"containers": []}
        if inst.cgroup_dir:
            try:
                with open(inst.cgroup_dir + "/cpu.stat", encoding="ascii") as fh:
                    for line in fh:
                        if line.startswith("usage_usec"):
                            stats["cpuUsageCoreNanoSeconds"] = (
                                int(line.split()[1]) * 1000)
                            break
                with open(inst.cgroup_dir + "/memory.current",
                          encoding="ascii") as fh:
                    stats["memoryUsageBytes"] = int(fh.read().strip())
            except OSError:
                pass
        tick_ns = 1_000_000_000 // os.sysconf("SC_CLK_TCK")
        page = os.sysconf("SC_PAGE_SIZE")
        cpu_total = 0
        mem_total = 0
        for c in inst.containers:
            entry: Dict[str, object] = {"name": c.name}
            try:
                with open(f"/proc/{c.pid}/stat", encoding="ascii") as fh:
                    fields = fh.read().rsplit(") ", 1)[-1].split()
                    # fields[11]=utime fields[12]=stime (post-comm offsets)
                    entry["cpuUsageCoreNanoSeconds"] = (
                        (int(fields[11]) + int(fields[12])) * tick_ns)
                with open(f"/proc/{c.pid}/statm", encoding="ascii") as fh:
                    entry["memoryRssBytes"] = int(fh.read().split()[1]) * page
                cpu_total += entry.get("cpuUsageCoreNanoSeconds", 0)
                mem_total += entry.get("memoryRssBytes", 0)
            except (OSError, IndexError, ValueError):
                pass  # container already exited
            stats["containers"].append(entry)
        stats.setdefault("cpuUsageCoreNanoSeconds", cpu_total)
        stats.setdefault("memoryUsageBytes", mem_total)
        return stats

    # ------------- logs -------------

    def get_log_path(self, instance_id: str, container: str = "") -> Optional[str]:
        """Filesystem path of a container's log (for `kubectl logs -f`
        streaming — the file is local, so follow is a tail)."""
        with self._lock:
            inst = self._instances.get(instance_id)
        if inst is None:
            return None
        names = ([c.name for c in inst.params.containers]
                 + [c.name for c in inst.params.init_containers]
                 + [c.name for c in inst.ephemeral_specs])
        if container and container in names:
            name = container
        else:
            name = (inst.params.containers[0].name
                    if inst.params.containers else "")
        return str(self.logs_dir / f"{instance_id}-{name}.log")

    def get_logs(self, instance_id: str, container: str = "",
                 tail: int = -1, previous: bool = False) -> str:
        """previous=True: the prior (pre-restart) run's log, like
        `kubectl logs --previous`."""
        path = self.get_log_path(instance_id, container)
        if path is None:
            return ""
        path = Path(path + ".prev") if previous else Path(path)
        if not path.exists():
            return ""
        text = path.read_text(errors="replace")
        if tail > 0:
            text = "\n".join(text.splitlines()[-tail:]) + "\n"
        return text

    def _run_confined(self, command: List[str], env: Dict[str, str],
                      cgroup_dir: str, uid: int, gid: int,
                      timeout_s: float, out_path: str = "",
                      setns_pid: int = -1, rootfs: str = "") -> int:
        """Run one command to completion inside the pod's confinement
        (cgroup + device filter + credentials — and for image pods, the
        container's own rootfs: setns into the live container's mount ns,
        or chroot into its rootfs copy) via the native launcher. Returns
        the exit code (127 = spawn failure, 124 = timeout killed). Used by
        kubectl-exec and by exec probes — k8s runs both INSIDE the
        container, never as the kubelet (root) on the host."""
        argv = list(command)
        if "/" not in argv[0]:
            # exec fast path does no PATH search — resolve against the
            # container's view when entering one, else the host
            if setns_pid >= 0:
                argv[0] = _resolve_via_proc_root(setns_pid, argv[0])
            elif rootfs:
                argv[0] = _resolve_in_tree(rootfs, argv[0])
            else:
                resolved = shutil.which(argv[0])
                if resolved:
                    argv[0] = resolved
        sink = out_path or "/dev/null"
        try:
            pid, pidfd, _, _, _ = self._native.launch_process(
                argv, [f"{k}={v}" for k, v in env.items()],
                "", sink, sink, cgroup_dir, True, False, uid, gid,
                False, "", rootfs, bool(rootfs), [], setns_pid,
            )
        except RuntimeError:
            return 127
        loop = self._native.EventLoop()
        loop.add_process(pid, pidfd, -1, 0)
        deadline = time.time() + timeout_s
        exit_code = None
        while time.time() < deadline and exit_code is None:
            for ev in loop.poll(100):
                if ev.type == "exited":
                    exit_code = ev.exit_code
        if exit_code is None:
            self._native.signal_process(pid, 9, True)
            exit_code = 124
        return exit_code

    def _container_entry(self, inst: Instance, cspec) -> tuple:
        """(setns_pid, rootfs) for entering the container of an image pod:
        mountns mode -> join the live container's namespaces; chroot mode
        -> its private rootfs copy; host pods -> (-1, "")."""
        if inst.image_mode == "mountns" and cspec is not None:
            cinfo = next((c for c in inst.containers
                          if c.name == cspec.name and c.exit_code is None
                          and c.pid > 0), None)
            if cinfo is None:
                cinfo = next((c for c in inst.containers
                              if c.exit_code is None and c.pid > 0), None)
            return (cinfo.pid if cinfo else -1), ""
        if inst.image_mode == "chroot" and cspec is not None \
                and self._rootfs_mgr is not None:
            rootfs = (self._rootfs_mgr.containers_dir
                      / f"{inst.id}-{cspec.name}" / "rootfs")
            if rootfs.is_dir():
                return -1, str(rootfs)
        return -1, ""

    def exec_in_instance(self, instance_id: str, command: List[str],
                         timeout_s: float = 30.0, container: str = "") -> tuple:
        """Non-interactive exec with the instance's environment (same GPU
        binding, same cgroup, the targeted container's credentials and —
        for image pods — its rootfs; kubectl exec -c selects `container`,
        default first). Returns (exit_code, combined_output)."""
        with self._lock:
            inst = self._instances.get(instance_id)
        if inst is None:
            return 127, f"instance {instance_id} not found"
        env = dict(os.environ)
        env.pop("ROCR_VISIBLE_DEVICES", None)
        env.pop("HIP_VISIBLE_DEVICES", None)
        env.update(inst.params.env)
        env.update(device_env(inst.gpu_indices, self.binder.ledger.inventory))
        cspec = None
        if container:
            cspec = next((c for c in inst.params.containers
                          if c.name == container), None)
            if cspec is None:
                return 127, f"container {container!r} not found in pod"
        elif inst.params.containers:
            cspec = inst.params.containers[0]
        uid = cspec.run_as_uid if cspec else -1
        gid = cspec.run_as_gid if cspec else -1
        setns_pid, rootfs = self._container_entry(inst, cspec)
        if inst.image_mode == "mountns" and setns_pid < 0:
            return 126, "no running container to exec into"
        out_path = self.logs_dir / f".exec-{inst.id}-{secrets.token_hex(4)}.log"
        exit_code = self._run_confined(
            list(command), env, inst.cgroup_dir, uid, gid, timeout_s,
            out_path=str(out_path), setns_pid=setns_pid, rootfs=rootfs)
        if exit_code == 127 and not out_path.exists():
            return 127, f"exec spawn failed: {command[0]!r}"
        output = out_path.read_text(errors="replace") if out_path.exists() else ""
        out_path.unlink(missing_ok=True)
        return exit_code, output


    def add_ephemeral_container(self, instance_id: str, cspec) -> None:
        """kubectl-debug ephemeral container: runs alongside the pod's
        containers (same cgroup, GPU env; for image pods it joins a live
        container's namespaces / rootfs copy) — never restarted, never
        gating readiness or completion. Raises on a dead pod or a
        duplicate name."""
        with self._lock:
            inst = self._instances.get(instance_id)
            if inst is None or inst.desired_status != PodStatus.RUNNING:
                raise RuntimeError(
                    f"instance {instance_id} is not running")
            all_names = ([c.name for c in inst.params.containers]
                         + [c.name for c in inst.params.init_containers]
                         + [c.name for c in inst.ephemeral_specs])
            if cspec.name in all_names:
                raise RuntimeError(
                    f"container name {cspec.name!r} already in use")
        params = inst.params
        env = dict(os.environ)
        env.pop("ROCR_VISIBLE_DEVICES", None)
        env.pop("HIP_VISIBLE_DEVICES", None)
        env.update(params.env)
        env.update(device_env(inst.gpu_indices, self.binder.ledger.inventory))
        env.update(cspec.env)
        env["AMDVK_INSTANCE_ID"] = inst.id
        argv = list(cspec.command) + list(cspec.args)
        if not argv:
            raise RuntimeError("ephemeral container needs a command")
        target = (inst.params.containers[0]
                  if inst.params.containers else None)
        setns_pid, rootfs = self._container_entry(inst, target)
        if "/" not in argv[0]:
            if setns_pid >= 0:
                argv[0] = _resolve_via_proc_root(setns_pid, argv[0])
            elif rootfs:
                argv[0] = _resolve_in_tree(rootfs, argv[0])
            else:
                resolved = shutil.which(argv[0])
                if resolved:
                    argv[0] = resolved
        stdout_path = str(self.logs_dir / f"{inst.id}-{cspec.name}.log")
        pid, pidfd, _, _, _ = self._native.launch_process(
            argv, [f"{k}={v}" for k, v in env.items()],
            cspec.working_dir or "", stdout_path, stdout_path,
            inst.cgroup_dir, True, False,
            cspec.run_as_uid, cspec.run_as_gid,
            False, "", rootfs, bool(rootfs), [], setns_pid,
        )
        cinfo = ContainerRuntimeInfo(name=cspec.name, pid=pid,
                                     started_at=time.time())
        with self._lock:
            inst.ephemeral_containers.append(cinfo)
            inst.ephemeral_specs.append(cspec)
            self._pid_to_instance[pid] = inst.id
        self._loop.add_process(pid, pidfd, -1, pid)
        self._persist(inst)
        self._notify(instance_id)
        log.info("ephemeral container started",
                 extra={"instance": instance_id, "container": cspec.name})

    # ------------- persistence / adoption -------------

    def _persist(self, inst: Instance) -> None:
        record = {
            "id": inst.id,
            "pod_key": inst.pod_key,
            "gpu_indices": inst.gpu_indices,
            "desired_status": inst.desired_status,
            "cgroup_dir": inst.cgroup_dir,
            "created_at": inst.created_at,
            "cost_per_hr": inst.cost_per_hr,
            "gpu_memory_bytes": inst.params.gpu_memory_bytes,
            "gpu_count": inst.params.gpu_count,
            "namespace": inst.params.namespace,
            "name": inst.params.name,
            "restart_policy": inst.params.restart_policy,
            # deadline/grace must survive restarts: an adopted RUNNING pod
            # keeps its activeDeadlineSeconds budget (timer re-armed with the
            # remainder in adopt_persisted) and its grace window
            "active_deadline_s": inst.params.active_deadline_s,
            "termination_grace_s": inst.params.termination_grace_s,
            "resolv_conf": inst.params.resolv_conf,
            # pod-level launch context: a crash-restart after kubelet
            # restart must rebuild the same mounts/identity/env
            "hostname": inst.params.hostname,
            "pod_env": inst.params.env,
            "fs_group": inst.params.fs_group,
            "host_aliases": [[ip, list(names)]
                             for ip, names in inst.params.host_aliases],
            "volumes": {n: dataclasses.asdict(v)
                        for n, v in inst.params.volumes.items()},
            "deadline_exceeded": inst.deadline_exceeded,
            "image_mode": inst.image_mode,
            "containers": [
                {
                    "name": c.name,
                    "pid": c.pid,
                    "started_at": c.started_at,
                    "finished_at": c.finished_at,
                    "exit_code": c.exit_code,
                    "ready": c.ready,
                    "restart_count": c.restart_count,
                    "image_id": c.image_id,
                }
                for c in inst.containers
            ],
            "ephemeral_containers": [
                {
                    "name": c.name,
                    "pid": c.pid,
                    "started_at": c.started_at,
                    "finished_at": c.finished_at,
                    "exit_code": c.exit_code,
                }
                for c in inst.ephemeral_containers
            ],
            "ephemeral_specs": [
                {"name": c.name, "image": c.image, "command": c.command,
                 "args": c.args, "env": c.env,
                 "run_as_uid": c.run_as_uid, "run_as_gid": c.run_as_gid}
                for c in inst.ephemeral_specs
            ],
            "init_containers": [
                {
                    "name": c.name,
                    "pid": c.pid,
                    "started_at": c.started_at,
                    "finished_at": c.finished_at,
                    "exit_code": c.exit_code,
                }
                for c in inst.init_containers
            ],
            "init_index": inst.init_index,
            "container_specs": [
                {
                    "name": c.name,
                    "image": c.image,
                    "command": c.command,
                    "args": c.args,
                    "tcp_ports": c.tcp_ports,
                    # probes must survive kubelet restarts: an adopted pod
                    # whose readinessProbe was lost could never become Ready
                    "readiness": dataclasses.asdict(c.readiness)
                    if c.readiness else None,
                    "liveness": dataclasses.asdict(c.liveness)
                    if c.liveness else None,
                    "startup": dataclasses.asdict(c.startup)
                    if c.startup else None,
                    "post_start": dataclasses.asdict(c.post_start)
                    if c.post_start else None,
                    "pre_stop": dataclasses.asdict(c.pre_stop)
                    if c.pre_stop else None,
                    "env": c.env,  # exec probes run in the container env
                    # credentials/cwd must survive restarts: an adopted pod
                    # whose container crash-restarts would otherwise relaunch
                    # as the kubelet's user (root) — silently dropping its
                    # securityContext.runAsUser
                    "run_as_uid": c.run_as_uid,
                    "run_as_gid": c.run_as_gid,
                    "run_as_non_root": c.run_as_non_root,
                    "read_only_root_fs": c.read_only_root_fs,
                    "termination_message_path": c.termination_message_path,
                    "termination_message_policy":
                        c.termination_message_policy,
                    "image_pull_policy": c.image_pull_policy,
                    "volume_mounts": [dataclasses.asdict(vm)
                                      for vm in c.volume_mounts],
                    "working_dir": c.working_dir,
                }
                for c in inst.params.containers
            ],
            "init_specs": [
                {
                    "name": c.name,
                    "image": c.image,
                    "command": c.command,
                    "args": c.args,
                    "run_as_uid": c.run_as_uid,
                    "run_as_gid": c.run_as_gid,
                    "working_dir": c.working_dir,
                }
                for c in inst.params.init_containers
            ],
        }
        # Unique temp name: the event thread and API threads may persist the
        # same instance concurrently; rename is atomic, last writer wins.
        tmp = self.instances_dir / f".{inst.id}.{threading.get_ident()}.tmp"
        tmp.write_text(json.dumps(record))
        try:
            tmp.rename(self.instances_dir / f"{inst.id}.json")
        except FileNotFoundError:
            pass  # state dir torn down during shutdown

    def adopt_persisted(self) -> List[str]:
        """Rebuild instance state after a kubelet restart: re-open pidfds for
        still-live processes, mark vanished ones EXITED, re-reserve GPUs.
        Returns adopted instance ids (reference LoadRunning analogue)."""
        adopted = []
        for path in sorted(self.instances_dir.glob("*.json")):
            try:
                rec = json.loads(path.read_text())
            except (OSError, json.JSONDecodeError):
                continue
            from .probes import ProbeSpec
            from .types import ContainerSpec, VolumeMount, VolumeSource

            params = DeployParams(
                pod_key=rec["pod_key"],
                name=rec.get("name", rec["pod_key"]),
                namespace=rec.get("namespace", "default"),
                gpu_count=rec.get("gpu_count", 0),
                gpu_memory_bytes=rec.get("gpu_memory_bytes", 0),
                restart_policy=rec.get("restart_policy", "Never"),
                active_deadline_s=rec.get("active_deadline_s", 0.0),
                resolv_conf=rec.get("resolv_conf", ""),
                hostname=rec.get("hostname", ""),
                env=rec.get("pod_env", {}) or {},
                fs_group=rec.get("fs_group", -1),
                host_aliases=[(ip, list(names)) for ip, names in
                              rec.get("host_aliases", []) or []],
                volumes={n: VolumeSource(**v) for n, v in
                         (rec.get("volumes", {}) or {}).items()},
                termination_grace_s=rec.get("termination_grace_s",
                                            TERM_GRACE_S),
                containers=[
                    ContainerSpec(
                        name=c["name"], image=c.get("image", ""),
                        command=c.get("command", []), args=c.get("args", []),
                        tcp_ports=c.get("tcp_ports", []),
                        env=c.get("env", {}) or {},
                        readiness=ProbeSpec(**c["readiness"])
                        if c.get("readiness") else None,
                        liveness=ProbeSpec(**c["liveness"])
                        if c.get("liveness") else None,
                        startup=ProbeSpec(**c["startup"])
                        if c.get("startup") else None,
                        post_start=ProbeSpec(**c["post_start"])
                        if c.get("post_start") else None,
                        pre_stop=ProbeSpec(**c["pre_stop"])
                        if c.get("pre_stop") else None,
                        run_as_uid=c.get("run_as_uid", -1),
                        run_as_gid=c.get("run_as_gid", -1),
                        run_as_non_root=c.get("run_as_non_root", False),
                        read_only_root_fs=c.get("read_only_root_fs",
                                                False),
                        termination_message_path=c.get(
                            "termination_message_path",
                            "/dev/termination-log"),
                        termination_message_policy=c.get(
                            "termination_message_policy", "File"),
                        image_pull_policy=c.get("image_pull_policy", ""),
                        volume_mounts=[VolumeMount(**vm) for vm in
                                       c.get("volume_mounts", []) or []],
                        working_dir=c.get("working_dir", ""),
                    )
                    for c in rec.get("container_specs", [])
                ],
                init_containers=[
                    ContainerSpec(
                        name=c["name"], image=c.get("image", ""),
                        command=c.get("command", []), args=c.get("args", []),
                        run_as_uid=c.get("run_as_uid", -1),
                        run_as_gid=c.get("run_as_gid", -1),
                        working_dir=c.get("working_dir", ""),
                    )
                    for c in rec.get("init_specs", [])
                ],
            )
            inst = Instance(
                id=rec["id"], pod_key=rec["pod_key"], params=params,
                gpu_indices=rec.get("gpu_indices", []),
                desired_status=rec.get("desired_status", PodStatus.RUNNING),
                cgroup_dir=rec.get("cgroup_dir", ""),
                created_at=rec.get("created_at", time.time()),
                cost_per_hr=rec.get("cost_per_hr", 0.0),
                init_index=rec.get("init_index", 0),
                deadline_exceeded=rec.get("deadline_exceeded", False),
                image_mode=rec.get("image_mode", ""),
            )
            for c in rec.get("init_containers", []):
                icinfo = ContainerRuntimeInfo(
                    name=c["name"], pid=c["pid"],
                    started_at=c.get("started_at", 0.0),
                    finished_at=c.get("finished_at", 0.0),
                    exit_code=c.get("exit_code"),
                )
                if icinfo.exit_code is None:
                    # Init was mid-flight across the restart: watch it again
                    # so progression resumes on its exit event.
                    pidfd = self._native.open_pidfd(icinfo.pid)
                    if pidfd >= 0:
                        self._loop.add_process(icinfo.pid, pidfd, -1, icinfo.pid)
                        with self._lock:
                            self._pid_to_instance[icinfo.pid] = inst.id
                    else:
                        icinfo.exit_code = -1
                        icinfo.finished_at = time.time()
                        icinfo.message = "init vanished during kubelet restart"
                inst.init_containers.append(icinfo)
            if any(c.exit_code not in (None, 0) for c in inst.init_containers):
                inst.desired_status = PodStatus.EXITED
            for c in rec.get("ephemeral_specs", []):
                inst.ephemeral_specs.append(ContainerSpec(
                    name=c["name"], image=c.get("image", ""),
                    command=c.get("command", []), args=c.get("args", []),
                    env=c.get("env", {}) or {},
                    run_as_uid=c.get("run_as_uid", -1),
                    run_as_gid=c.get("run_as_gid", -1)))
            for c in rec.get("ephemeral_containers", []):
                einfo = ContainerRuntimeInfo(
                    name=c["name"], pid=c["pid"],
                    started_at=c.get("started_at", 0.0),
                    finished_at=c.get("finished_at", 0.0),
                    exit_code=c.get("exit_code"))
                if einfo.exit_code is None:
                    pidfd = self._native.open_pidfd(einfo.pid)
                    if pidfd >= 0:
                        self._loop.add_process(einfo.pid, pidfd, -1,
                                               einfo.pid)
                        with self._lock:
                            self._pid_to_instance[einfo.pid] = inst.id
                    else:
                        einfo.exit_code = -1
                        einfo.finished_at = time.time()
                inst.ephemeral_containers.append(einfo)
            all_alive = True
            for c in rec.get("containers", []):
                cinfo = ContainerRuntimeInfo(
                    name=c["name"], pid=c["pid"], started_at=c.get("started_at", 0.0),
                    finished_at=c.get("finished_at", 0.0),
                    exit_code=c.get("exit_code"), ready=c.get("ready", False),
                    restart_count=c.get("restart_count", 0),
                    image_id=c.get("image_id", ""),
                )
                if cinfo.exit_code is None:
                    pidfd = self._native.open_pidfd(cinfo.pid)
                    if pidfd >= 0:
                        self._loop.add_process(cinfo.pid, pidfd, -1, cinfo.pid)
                        with self._lock:
                            self._pid_to_instance[cinfo.pid] = inst.id
                        if not cinfo.ready:
                            # The AMDVK_READY_FD pipe died with the old
                            # kubelet: its signal can never arrive now. A
                            # container WITH a readinessProbe re-proves
                            # readiness via the probe loop; without one,
                            # k8s semantics apply — a running container is
                            # ready (found by soak race hunting: a pod
                            # deployed moments before a kubelet crash was
                            # stuck NotReady forever).
                            spec = next(
                                (s for s in params.containers
                                 if s.name == cinfo.name), None)
                            if spec is None or spec.readiness is None:
                                cinfo.ready = True
                    else:
                        # Process died while we were away; exact code unknown.
                        cinfo.exit_code = -1
                        cinfo.finished_at = time.time()
                        cinfo.message = "process vanished during kubelet restart"
                        all_alive = False
                inst.containers.append(cinfo)
            # Containers that crashed across the restart window resume their
            # restartPolicy loop instead of failing the pod: re-reserve the
            # GPUs and schedule restarts with fresh backoff.
            def wants_restart(c) -> bool:
                if inst.desired_status != PodStatus.RUNNING:
                    return False
                if params.restart_policy == "Always":
                    return c.exit_code is not None
                if params.restart_policy == "OnFailure":
                    return c.exit_code is not None and c.exit_code != 0
                return False

            restartable = [c for c in inst.containers if wants_restart(c)]
            all_exited = inst.containers and all(
                c.exit_code is not None for c in inst.containers)
            if all_exited and not restartable:
                if inst.desired_status not in (PodStatus.TERMINATED,):
                    inst.desired_status = PodStatus.EXITED
            elif inst.desired_status == PodStatus.RUNNING and inst.gpu_indices:
                # Live (or about-to-restart) pod keeps owning its GPUs.
                self.ledger.adopt(
                    inst.pod_key, inst.gpu_indices,
                    BindRequest(inst.pod_key, len(inst.gpu_indices),
                                params.gpu_memory_bytes).bytes_per_gpu,
                )
            if inst.cgroup_dir:
                # Keep the slot counter ahead of adopted slots so fresh pods
                # never collide with a cgroup still owned by a live pod.
                base = os.path.basename(inst.cgroup_dir)
                if base.startswith("slot"):
                    try:
                        with self._lock:
                            self._cgroup_counter = max(
                                self._cgroup_counter, int(base[4:])
                            )
                    except ValueError:
                        pass
            with self._lock:
                self._instances[inst.id] = inst
                for c in restartable:
                    self._schedule_restart(inst, c)
                if (params.active_deadline_s > 0 and not inst.deadline_exceeded
                        and inst.desired_status in (PodStatus.RUNNING,
                                                    PodStatus.STARTING)):
                    # Re-arm activeDeadlineSeconds with the *remaining*
                    # budget (deadline anchored at deploy time, not at
                    # adoption); an already-blown budget fires immediately.
                    remaining = (inst.created_at + params.active_deadline_s
                                 - time.time())
                    t = threading.Timer(max(0.0, remaining),
                                        self._deadline_exceeded,
                                        args=(inst.id,))
                    t.daemon = True
                    self._deadline_timers[inst.id] = t
                    t.start()
                if inst.desired_status == PodStatus.TERMINATING and any(
                        c.exit_code is None for c in inst.containers):
                    # the previous kubelet died mid-grace: its SIGKILL
                    # timer died with it — re-arm with the full grace
                    # (elapsed time is unknown; a TERM-immune pid-1 must
                    # not outlive the window forever)
                    t = threading.Timer(
                        max(0.1, params.termination_grace_s),
                        self._force_kill, args=(inst.id,))
                    t.daemon = True
                    self._kill_timers[inst.id] = t
                    t.start()
            adopted.append(inst.id)
        if adopted:
            log.info("adopted persisted instances", extra={"count": len(adopted)})
        return adopted

    def close(self) -> None:
        self._stop.set()
        self._probe_ticker.stop()
        self._logrotate_ticker.stop()
        self._loop.wake()
        self._watcher.join(timeout=2.0)
        for timer in self._kill_timers.values():
            timer.cancel()
        for timer in self._restart_timers.values():
            timer.cancel()
        for timer in self._deadline_timers.values():
            timer.cancel()


def _resolve_via_proc_root(pid: int, argv0: str) -> str:
    """PATH-resolve inside a live container via /proc/<pid>/root (the
    kernel's view of its mount namespace root)."""
    for p in ("/usr/local/sbin", "/usr/local/bin", "/usr/sbin", "/usr/bin",
              "/sbin", "/bin"):
        if os.path.exists(f"/proc/{pid}/root{p}/{argv0}"):
            return f"{p}/{argv0}"
    return argv0


def _resolve_in_tree(rootfs: str, argv0: str) -> str:
    """PATH-resolve against a rootfs directory tree (chroot mode)."""
    for p in ("/usr/local/sbin", "/usr/local/bin", "/usr/sbin", "/usr/bin",
              "/sbin", "/bin"):
        if os.path.exists(f"{rootfs}{p}/{argv0}"):
            return f"{p}/{argv0}"
    return argv0


def _listening_tcp_inodes() -> dict:
    """socket-inode → local port for LISTEN-state TCP sockets from
    /proc/net/tcp{,6} (state 0A, inode field 9). Two passes, unioned:
    the kernel's seq_file iteration can skip entries while the socket
    table mutates under load, and a missed entry here reads as
    "pod port not exposed"."""
    by_inode = {}
    for path in ("/proc/net/tcp", "/proc/net/tcp6",
                 "/proc/net/tcp", "/proc/net/tcp6"):
        try:
            with open(path, "r", encoding="ascii") as fh:
                next(fh, None)
                for line in fh:
                    parts = line.split()
                    if len(parts) > 9 and parts[3] == "0A":
                        try:
                            port = int(parts[1].rsplit(":", 1)[1], 16)
                            by_inode[int(parts[9])] = port
                        except ValueError:
                            continue
        except OSError:
            continue
    return by_inode
