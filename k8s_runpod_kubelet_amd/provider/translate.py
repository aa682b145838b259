"""Spec translation: Kubernetes pod → local deploy parameters.

Counterpart of the reference's translation layer
(pkg/virtual_kubelet/runpod_client.go:1023-1393):

- owner-Job resolution + annotation fallback (getOwnerJob :1056-1099,
  getAnnotationWithFallback :1101-1112),
- cloud-type validation (validateCloudType :1114-1134 — SECURE default,
  SECURE/COMMUNITY accepted; locally SECURE means "cgroup-isolated", the only
  mode implemented, COMMUNITY is accepted and treated the same),
- datacenter-ID compliance against the node-level allow-list
  (validateDatacenterIDs :1136-1178 — kept verbatim: the "datacenters" of a
  single node deployment are its configured partition labels),
- GPU memory annotation parse with the 16 GB default (extractGPUMemory
  :1180-1191),
- port extraction (ports.py), env extraction (envvars.py),
- parameter assembly (PrepareRunPodParameters :1248-1377) — but targeting the
  local binder: gpu_count comes from the ``amd.com/gpu`` resource request
  (the reference picks a cloud GPU *type*; here the type is always MI355X and
  the choice is *which* GPUs, made by the binder at deploy time).
"""

from __future__ import annotations

import logging
from typing import Any, Dict, List, Optional

from ..config import Config
from ..kube.client import K8sClient, is_not_found
from ..kube.objects import (
    namespace_of,
    owner_references,
    pod_key,
    resource_parse_bytes,
    resource_parse_cpu,
)
from ..runtime.probes import ProbeSpec
from ..runtime.types import (ContainerSpec, DeployParams, VolumeMount,
                             VolumeSource)
from . import annotations as ann
from .envvars import extract_env_vars
from .ports import get_requested_ports, tcp_ports_of_container


def _volume_mounts_of(c: Dict[str, Any]) -> List[VolumeMount]:
    out: List[VolumeMount] = []
    for m in c.get("volumeMounts", []) or []:
        if m.get("name") and m.get("mountPath"):
            out.append(VolumeMount(
                name=m["name"], mount_path=m["mountPath"],
                read_only=bool(m.get("readOnly")),
                sub_path=m.get("subPath", "") or ""))
    return out


def extract_volumes(pod: Dict[str, Any], client: K8sClient) -> Dict[str, VolumeSource]:
    """spec.volumes -> resolved VolumeSource map (consumed by image-backed
    containers; host-process pods see the host filesystem, and secret
    volumes are additionally flattened into env for reference parity —
    runpod_client.go:949-979). Secret/ConfigMap content is fetched at
    translation time like env extraction; downwardAPI items project pod
    metadata; projected volumes merge those three source kinds (plus a
    best-effort serviceAccountToken) into one directory.
    Unsupported volume types (PVC, CSI, ...) are
    skipped — mounts referencing them are ignored rather than failing the
    pod."""
    from .envvars import ConfigMapCollector, SecretCollector

    namespace = pod.get("metadata", {}).get("namespace", "default")
    out: Dict[str, VolumeSource] = {}
    for vol in pod.get("spec", {}).get("volumes", []) or []:
        name = vol.get("name", "")
        if not name:
            continue
        if "emptyDir" in vol:
            ed = vol.get("emptyDir") or {}
            size = 0
            if ed.get("sizeLimit"):
                size = int(resource_parse_bytes(str(ed["sizeLimit"])))
            out[name] = VolumeSource(kind="emptyDir",
                                     medium=str(ed.get("medium", "") or ""),
                                     size_limit_bytes=size)
        elif (vol.get("hostPath") or {}).get("path"):
            out[name] = VolumeSource(kind="hostPath",
                                     host_path=vol["hostPath"]["path"])
        elif (vol.get("secret") or {}).get("secretName"):
            sec = vol["secret"]
            col = SecretCollector(client, namespace)
            data = col.get(sec["secretName"])
            if not col.exists(sec["secretName"]) and \
                    not bool(sec.get("optional")):
                # k8s: a non-optional missing secret blocks the pod
                # (ContainerCreating); Pending-with-retry is the local
                # equivalent — the pod starts once the secret appears
                raise ValidationError(
                    f'volume "{name}": secret '
                    f'{sec["secretName"]!r} not found')
            files = _project_items(data, sec.get("items"))
            out[name] = VolumeSource(kind="files", files=files,
                                     file_mode=int(sec.get("defaultMode",
                                                           0o644)))
        elif (vol.get("configMap") or {}).get("name"):
            cm = vol["configMap"]
            col = ConfigMapCollector(client, namespace)
            data = col.get(cm["name"])
            if not col.exists(cm["name"]) and not bool(cm.get("optional")):
                raise ValidationError(
                    f'volume "{name}": configMap {cm["name"]!r} not found')
            files = _project_items(data, cm.get("items"))
            out[name] = VolumeSource(kind="files", files=files,
                                     file_mode=int(cm.get("defaultMode",
                                                          0o644)))
        elif vol.get("downwardAPI") is not None:
            da = vol["downwardAPI"] or {}
            files = {}
            for item in da.get("items", []) or []:
                path = item.get("path", "")
                fref = (item.get("fieldRef") or {}).get("fieldPath", "")
                if path and fref:
                    files[path] = _downward_field(pod, fref)
            out[name] = VolumeSource(kind="files", files=files,
                                     file_mode=int(da.get("defaultMode",
                                                          0o644)))
        elif vol.get("projected") is not None:
            pr = vol["projected"] or {}
            files: Dict[str, str] = {}
            for src in pr.get("sources", []) or []:
                if (src.get("secret") or {}).get("name"):
                    s = src["secret"]
                    data = SecretCollector(client, namespace).get(s["name"])
                    files.update(_project_items(data, s.get("items")))
                elif (src.get("configMap") or {}).get("name"):
                    c = src["configMap"]
                    data = ConfigMapCollector(client, namespace).get(c["name"])
                    files.update(_project_items(data, c.get("items")))
                elif src.get("downwardAPI") is not None:
                    for item in (src["downwardAPI"] or {}).get("items",
                                                               []) or []:
                        path = item.get("path", "")
                        fref = (item.get("fieldRef") or {}).get("fieldPath",
                                                                "")
                        if path and fref:
                            files[path] = _downward_field(pod, fref)
                elif (src.get("serviceAccountToken") or {}).get("path"):
                    # no token issuer offline: project the kubelet's own SA
                    # token if one is mounted, else an empty placeholder so
                    # workloads see the expected file
                    files[src["serviceAccountToken"]["path"]] = \
                        _local_sa_token()
            out[name] = VolumeSource(kind="files", files=files,
                                     file_mode=int(pr.get("defaultMode",
                                                          0o644)))
    return out


def _local_sa_token() -> str:
    try:
        with open("/var/run/secrets/kubernetes.io/serviceaccount/token",
                  "r", encoding="utf-8") as fh:
            return fh.read()
    except OSError:
        return ""


def _downward_field(pod: Dict[str, Any], path: str) -> str:
    """Downward-API fieldRef subset for volume projection (same fields as
    the env fieldRef support in envvars.py, plus labels/annotations)."""
    md = pod.get("metadata", {})
    if path == "metadata.labels":
        return "\n".join(f'{k}="{v}"'
                          for k, v in sorted((md.get("labels") or {}).items()))
    if path == "metadata.annotations":
        return "\n".join(f'{k}="{v}"'
                          for k, v in sorted((md.get("annotations")
                                              or {}).items()))
    return {
        "metadata.name": md.get("name", ""),
        "metadata.namespace": md.get("namespace", "default"),
        "metadata.uid": md.get("uid", ""),
        "spec.nodeName": pod.get("spec", {}).get("nodeName", ""),
    }.get(path, "")


def _project_items(data: Dict[str, str], items) -> Dict[str, str]:
    """k8s items[] projection: each item's key lands at its path (default:
    key name)."""
    if not items:
        return dict(data)
    files: Dict[str, str] = {}
    for item in items:
        key = item.get("key", "")
        if key in data:
            files[item.get("path") or key] = data[key]
    return files


log = logging.getLogger("provider.translate")

GIB = 1024**3


class ValidationError(ValueError):
    pass


def get_owner_job(pod: Dict[str, Any], client: K8sClient) -> Optional[Dict[str, Any]]:
    """Walk ownerReferences for Kind=Job, fetch it, verify UID
    (runpod_client.go:1056-1099)."""
    for ref in owner_references(pod):
        if ref.get("kind") != "Job":
            continue
        try:
            job = client.get_job(namespace_of(pod), ref.get("name", ""))
        except Exception as exc:
            if is_not_found(exc):
                continue
            raise
        if job.get("metadata", {}).get("uid") == ref.get("uid"):
            return job
    return None


def annotation_with_fallback(
    pod: Dict[str, Any], job: Optional[Dict[str, Any]], key: str, default: str = ""
) -> str:
    """pod annotation → owner-Job annotation → default
    (runpod_client.go:1101-1112)."""
    value = pod.get("metadata", {}).get("annotations", {}).get(key, "")
    if value:
        return value
    if job is not None:
        value = job.get("metadata", {}).get("annotations", {}).get(key, "")
        if value:
            return value
    return default


def validate_cloud_type(value: str) -> str:
    """Default SECURE; normalize upper; only SECURE/COMMUNITY accepted
    (runpod_client.go:1114-1134)."""
    if not value:
        return "SECURE"
    norm = value.upper()
    if norm not in ("SECURE", "COMMUNITY"):
        raise ValidationError(f"invalid cloud type {value!r}: must be SECURE or COMMUNITY")
    return norm


def validate_datacenter_ids(pod_value: str, node_allow_list: List[str]) -> List[str]:
    """Node-level allow-list semantics (runpod_client.go:1136-1178): pod list
    filtered to the subset allowed by the node config; empty intersection is a
    hard error; no node config ⇒ pod value passes through."""
    pod_ids = [s.strip() for s in pod_value.split(",") if s.strip()] if pod_value else []
    if not node_allow_list:
        return pod_ids
    if not pod_ids:
        return list(node_allow_list)
    allowed = [d for d in pod_ids if d in node_allow_list]
    if not allowed:
        raise ValidationError(
            f"pod datacenter ids {pod_ids} not allowed by node config {node_allow_list}"
        )
    return allowed


def extract_gpu_memory_gb(pod: Dict[str, Any], job: Optional[Dict[str, Any]],
                          default_gb: int) -> int:
    """Annotation string → int GiB, default 16 (runpod_client.go:1180-1191).
    Accepts both runpod.io/required-gpu-memory and runpod.io/gpu-memory, and
    tolerates quantity suffixes ("256GiB")."""
    raw = annotation_with_fallback(pod, job, ann.GPU_MEMORY) or annotation_with_fallback(
        pod, job, ann.GPU_MEMORY_ALT
    )
    if not raw:
        return default_gb
    text = raw.strip()
    for suffix in ("GiB", "Gi", "GB", "G"):
        if text.endswith(suffix):
            text = text[: -len(suffix)]
            break
    try:
        # OverflowError covers float('inf')-family strings ("INF", "1e999")
        return max(0, int(float(text)))
    except (ValueError, OverflowError):
        log.warning("unparseable gpu-memory annotation", extra={"value": raw})
        return default_gb


def gpu_count_of(pod: Dict[str, Any]) -> int:
    """amd.com/gpu from resource limits/requests across containers."""
    total = 0
    for container in pod.get("spec", {}).get("containers", []) or []:
        res = container.get("resources", {}) or {}
        for section in ("limits", "requests"):
            value = (res.get(section) or {}).get(ann.GPU_RESOURCE)
            if value is not None:
                try:
                    total += int(value)
                except (TypeError, ValueError):
                    raise ValidationError(
                        f"invalid {ann.GPU_RESOURCE} quantity: {value!r}")
                break
    return total


def _int_or_error(value, what: str) -> int:
    try:
        return int(value)
    except (TypeError, ValueError):
        raise ValidationError(f"invalid {what}: {value!r}")


def _float_or(value, default: float) -> float:
    try:
        return float(value)
    except (TypeError, ValueError):
        return default


def _render_dns_config(pod: Dict[str, Any]) -> str:
    """spec.dnsConfig -> resolv.conf text ("" = inherit the node's file,
    dnsPolicy Default semantics — there is no cluster DNS on a
    single-node deployment, so ClusterFirst falls back to the node too).
    nameservers/searches/options follow the kubelet's rendering."""
    dc = pod.get("spec", {}).get("dnsConfig") or {}
    if not dc:
        return ""
    lines = []
    for ns in dc.get("nameservers", []) or []:
        lines.append(f"nameserver {ns}")
    searches = [s for s in (dc.get("searches", []) or []) if s]
    if searches:
        lines.append("search " + " ".join(searches))
    opts = []
    for o in dc.get("options", []) or []:
        name = o.get("name", "")
        if not name:
            continue
        val = o.get("value")
        opts.append(f"{name}:{val}" if val not in (None, "") else name)
    if opts:
        lines.append("options " + " ".join(opts))
    return ("\n".join(lines) + "\n") if lines else ""


def _cgroup_limits(pod: Dict[str, Any]) -> (str, str):
    """Aggregate container CPU/memory limits into cgroup v2 strings; a
    malformed quantity is a ValidationError (the apiserver would normally
    reject it; a fake/relaxed one must not crash translation)."""
    cpu_cores = 0.0
    mem_bytes = 0
    for container in pod.get("spec", {}).get("containers", []) or []:
        limits = (container.get("resources", {}) or {}).get("limits") or {}
        try:
            if "cpu" in limits:
                cpu_cores += resource_parse_cpu(limits["cpu"])
            if "memory" in limits:
                mem_bytes += resource_parse_bytes(limits["memory"])
        except (TypeError, ValueError):
            raise ValidationError(
                f"invalid resource quantity in limits: {limits!r}")
    cpu_max = f"{int(cpu_cores * 100000)} 100000" if cpu_cores > 0 else ""
    memory_max = str(mem_bytes) if mem_bytes > 0 else ""
    return cpu_max, memory_max


def prepare_deploy_params(
    pod: Dict[str, Any],
    client: K8sClient,
    config: Config,
    gpu_offers=None,  # selector.GpuOfferCatalog | None — fail-fast feasibility
) -> DeployParams:
    """PrepareRunPodParameters analogue (runpod_client.go:1248-1377)."""
    metadata = pod.get("metadata", {})
    job = get_owner_job(pod, client)

    cloud_type = validate_cloud_type(
        annotation_with_fallback(pod, job, ann.CLOUD_TYPE)
    )
    datacenters = validate_datacenter_ids(
        annotation_with_fallback(pod, job, ann.DATACENTER_IDS), config.datacenter_ids
    )
    template_id = annotation_with_fallback(pod, job, ann.TEMPLATE_ID)
    registry_auth = annotation_with_fallback(pod, job, ann.REGISTRY_AUTH_ID)

    gpu_count = gpu_count_of(pod)
    gpu_memory_gb = extract_gpu_memory_gb(pod, job, config.gpu_memory_default_gb)
    # A pod that names a GPU memory requirement but no resource count gets one
    # GPU (the reference deploys gpuCount 1 implicitly via the type selector).
    if gpu_count == 0 and (
        ann.GPU_MEMORY in metadata.get("annotations", {})
        or ann.GPU_MEMORY_ALT in metadata.get("annotations", {})
    ):
        gpu_count = 1

    # Fail-fast feasibility check — the GetGPUTypes analogue
    # (runpod_client.go:1281: GetGPUTypes(minRAM, maxPrice, cloudType) errors
    # when no type matches). max_gpu_price is wired (dead in the reference).
    if gpu_count > 0 and gpu_offers is not None:
        offers = gpu_offers.offers(
            min_memory_bytes=(gpu_memory_gb * GIB + gpu_count - 1) // max(gpu_count, 1),
            max_cost=config.max_gpu_price,
        )
        if len(offers) < gpu_count:
            raise ValidationError(
                f"no GPU set available: need {gpu_count} × "
                f"{gpu_memory_gb // max(gpu_count, 1)} GiB headroom at cost <= "
                f"{config.max_gpu_price}; {len(offers)} GPUs eligible"
            )

    pod_env, per_container_env = extract_env_vars(pod, client)
    # ports override with owner-Job fallback (runpod_client.go:1310-1330)
    ports_override = annotation_with_fallback(pod, job, ann.PORTS)
    if ports_override:
        requested_ports = [p.strip() for p in ports_override.split(",") if p.strip()]
    else:
        requested_ports = get_requested_ports(pod)

    # securityContext.runAsUser/runAsGroup: pod-level default, container-level
    # override (the reference's containers run as their image's user; local
    # processes honor the same kubectl-facing fields).
    pod_sc = pod.get("spec", {}).get("securityContext", {}) or {}
    containers: List[ContainerSpec] = []
    for i, c in enumerate(pod.get("spec", {}).get("containers", []) or []):
        c_sc = c.get("securityContext", {}) or {}
        uid = c_sc.get("runAsUser", pod_sc.get("runAsUser"))
        gid = c_sc.get("runAsGroup", pod_sc.get("runAsGroup"))
        non_root = bool(c_sc.get("runAsNonRoot",
                                 pod_sc.get("runAsNonRoot", False)))
        containers.append(
            ContainerSpec(
                name=c.get("name", f"c{i}"),
                image=c.get("image", ""),
                command=list(c.get("command", []) or []),
                args=list(c.get("args", []) or []),
                env=per_container_env[i] if i < len(per_container_env) else {},
                working_dir=c.get("workingDir", ""),
                tcp_ports=tcp_ports_of_container(c),
                run_as_uid=(_int_or_error(uid, "runAsUser")
                            if uid not in (None, "") else -1),
                run_as_gid=(_int_or_error(gid, "runAsGroup")
                            if gid not in (None, "") else -1),
                liveness=ProbeSpec.parse(c.get("livenessProbe")),
                readiness=ProbeSpec.parse(c.get("readinessProbe")),
                startup=ProbeSpec.parse(c.get("startupProbe")),
                post_start=ProbeSpec.parse_hook(
                    (c.get("lifecycle") or {}).get("postStart")),
                pre_stop=ProbeSpec.parse_hook(
                    (c.get("lifecycle") or {}).get("preStop")),
                termination_message_path=c.get(
                    "terminationMessagePath", "/dev/termination-log")
                or "/dev/termination-log",
                termination_message_policy=c.get(
                    "terminationMessagePolicy", "File") or "File",
                image_pull_policy=c.get("imagePullPolicy", "") or "",
                run_as_non_root=non_root,
                read_only_root_fs=bool(
                    c_sc.get("readOnlyRootFilesystem", False)),
                volume_mounts=_volume_mounts_of(c),
            )
        )
    if not containers:
        raise ValidationError("pod has no containers")

    # spec.initContainers (sequential, to completion, before mains) — real
    # kubelet behavior the reference ignores entirely (it reads only
    # Containers[0], runpod_client.go:1028-1030).
    init_containers: List[ContainerSpec] = []
    for i, c in enumerate(pod.get("spec", {}).get("initContainers", []) or []):
        c_sc = c.get("securityContext", {}) or {}
        uid = c_sc.get("runAsUser", pod_sc.get("runAsUser"))
        gid = c_sc.get("runAsGroup", pod_sc.get("runAsGroup"))
        non_root = bool(c_sc.get("runAsNonRoot",
                                 pod_sc.get("runAsNonRoot", False)))
        init_containers.append(
            ContainerSpec(
                name=c.get("name", f"init{i}"),
                image=c.get("image", ""),
                command=list(c.get("command", []) or []),
                args=list(c.get("args", []) or []),
                env=dict(pod_env),
                working_dir=c.get("workingDir", ""),
                run_as_uid=(_int_or_error(uid, "runAsUser")
                            if uid not in (None, "") else -1),
                run_as_gid=(_int_or_error(gid, "runAsGroup")
                            if gid not in (None, "") else -1),
                volume_mounts=_volume_mounts_of(c),
            )
        )

    cpu_max, memory_max = _cgroup_limits(pod)
    resolv = _render_dns_config(pod)

    return DeployParams(
        pod_key=pod_key(pod),
        name=metadata.get("name", ""),
        namespace=metadata.get("namespace", "default"),
        containers=containers,
        init_containers=init_containers,
        env=pod_env,
        gpu_count=gpu_count,
        gpu_memory_bytes=gpu_memory_gb * GIB,
        max_gpu_cost=config.max_gpu_price,
        requested_ports=requested_ports,
        cloud_type=cloud_type,
        datacenter_ids=datacenters,
        template_id=template_id,
        registry_auth_id=registry_auth,
        cpu_limit=cpu_max,
        resolv_conf=resolv,
        memory_limit=memory_max,
        volumes=extract_volumes(pod, client),
        fs_group=_int_or_error(pod_sc.get("fsGroup"), "fsGroup")
        if pod_sc.get("fsGroup") not in (None, "") else -1,
        host_aliases=[
            (ha.get("ip", ""), list(ha.get("hostnames", []) or []))
            for ha in pod.get("spec", {}).get("hostAliases", []) or []
            if ha.get("ip") and ha.get("hostnames")
        ],
        labels=dict(metadata.get("labels", {}) or {}),
        hostname=pod.get("spec", {}).get("hostname")
        or metadata.get("name", ""),
        # k8s semantics: a pod without restartPolicy means Always (apiserver
        # admission normally sets it before the kubelet sees the pod; this
        # default covers pods arriving un-defaulted). Operators who want the
        # reference's run-to-completion model (a stopped RunPod instance is
        # EXITED/Succeeded, kubelet.go:1906) set
        # config.restart_policy_default: Never.
        restart_policy=pod.get("spec", {}).get("restartPolicy")
        or config.restart_policy_default,
        termination_grace_s=_float_or(
            pod.get("spec", {}).get("terminationGracePeriodSeconds", 30), 30.0),
        active_deadline_s=_float_or(
            pod.get("spec", {}).get("activeDeadlineSeconds", 0) or 0, 0.0),
    )
