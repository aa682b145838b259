"""CLI entry point.

Flag surface kept compatible with the reference's 13 flags
(cmd/virtual_kubelet/main.go:59-73) plus the local-backend knobs; dead
reference flags are wired here (``--max-gpu-price`` reaches the selector,
``--log-level`` is applied, ``--reconcile-interval`` drives the real loops —
SURVEY §5.6 inconsistency list)."""

from __future__ import annotations

import argparse
import logging
import os
import sys

from .app import build_stack, run_forever
from .config import load_config
from .logging_setup import initialize_logger
from .utils.backoff import parse_duration_s

log = logging.getLogger("cli")


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="amd-virtual-kubelet",
        description="MI355X-native virtual-kubelet provider",
    )
    # reference flag surface (main.go:59-73)
    p.add_argument("--kubeconfig", default="", help="path to kubeconfig")
    p.add_argument("--reconcile-interval", default="30s",
                   help="status reconcile interval (also informer resync)")
    p.add_argument("--max-gpu-price", type=float, default=0.5,
                   help="max per-GPU occupancy cost in [0,1] (price remap)")
    p.add_argument("--health-server-address", default=":8080")
    p.add_argument("--nodename", default="virtual-runpod")
    p.add_argument("--os", dest="operating_system", default="Linux")
    p.add_argument("--provider-config", default="", help="YAML config file")
    p.add_argument("--internal-ip", default="127.0.0.1")
    p.add_argument("--listen-port", type=int, default=10250)
    p.add_argument("--log-level", default="info")
    p.add_argument("--namespace", default="kube-system")
    p.add_argument("--datacenter-ids", default="",
                   help="comma-separated allow-list")
    p.add_argument("--heartbeat-interval", default="300s")
    # local backend knobs
    p.add_argument("--state-dir", default="")
    p.add_argument("--sysfs-root", default="/sys")
    p.add_argument("--runtime", choices=["process", "fake"], default="process")
    p.add_argument("--workers", type=int, default=4,
                   help="pod controller workers (reference uses 1)")
    p.add_argument("--image-store-dir", default="",
                   help="node-local OCI image store (default: "
                        "<state-dir>/images)")
    p.add_argument("--image-isolation", choices=["auto", "mountns", "chroot"],
                   default="auto",
                   help="image rootfs isolation: overlay+mount-ns+pivot_root "
                        "(mountns), chroot fallback, or probe (auto)")
    p.add_argument("--image-registry", default="",
                   help="in-cluster registry base URL; images missing from "
                        "the local store are pulled from here at deploy "
                        "time (token via AMDVK_REGISTRY_TOKEN)")
    p.add_argument("--gpu-count-override", type=int, default=-1,
                   help="force synthetic GPU inventory (CPU-only dev)")
    p.add_argument("--fake-apiserver", action="store_true",
                   help="start an in-process fake Kubernetes apiserver and "
                        "connect to it (self-contained demo; no cluster "
                        "needed — create pods with curl/kubectl against the "
                        "printed URL)")
    return p


def validate_environment() -> None:
    """Reference validateEnvironment hard-exits without RUNPOD_API_KEY
    (main.go:306-311); the local backend needs no API key — instead warn when
    neither /dev/kfd nor a synthetic override is present."""
    if not os.path.exists("/dev/kfd"):
        log.warning("no /dev/kfd on this host; GPU inventory will be synthetic")


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    initialize_logger(args.log_level, os.environ.get("AMDVK_JSON_LOG") or None)

    cfg = load_config(args.provider_config or None)
    cfg.node_name = args.nodename
    cfg.operating_system = args.operating_system
    cfg.internal_ip = args.internal_ip
    cfg.listen_port = args.listen_port
    cfg.namespace = args.namespace
    cfg.kubeconfig = args.kubeconfig
    cfg.log_level = args.log_level
    cfg.max_gpu_price = args.max_gpu_price
    cfg.reconcile_interval_s = parse_duration_s(args.reconcile_interval, 30.0)
    cfg.heartbeat_interval_s = parse_duration_s(args.heartbeat_interval, 300.0)
    if args.datacenter_ids:
        cfg.datacenter_ids = [s.strip() for s in args.datacenter_ids.split(",") if s.strip()]
    if args.state_dir:
        cfg.state_dir = args.state_dir
    cfg.sysfs_root = args.sysfs_root
    cfg.runtime = args.runtime
    cfg.pod_controller_workers = args.workers
    cfg.gpu_count_override = args.gpu_count_override
    if args.image_store_dir:
        cfg.image_store_dir = args.image_store_dir
    cfg.image_isolation = args.image_isolation
    if args.image_registry:
        cfg.image_registry = args.image_registry

    validate_environment()
    fake_srv = None
    client = None
    if args.fake_apiserver:
        from .kube.fake_apiserver import FakeApiServer
        from .kube.real import ClusterConfig, HttpK8sClient

        fake_srv = FakeApiServer().start()
        client = HttpK8sClient(ClusterConfig(server=fake_srv.url))
        print(f"fake apiserver at {fake_srv.url} — e.g.\n"
              f"  curl -s {fake_srv.url}/api/v1/nodes/{cfg.node_name} | head\n"
              f"  curl -s -XPOST {fake_srv.url}/api/v1/namespaces/default/pods"
              f" -d @pod.json")
    stack = build_stack(cfg, client=client)
    # Auth introspection (reference logAuthInfo, main.go:92-108).
    try:
        review = stack.client.self_subject_review()
        user = review.get("status", {}).get("userInfo", {})
        if user:
            log.info("authenticated", extra={"user": user.get("username"),
                                             "groups": user.get("groups")})
    except Exception:
        log.debug("self subject review unavailable")
    run_forever(stack)
    return 0


if __name__ == "__main__":
    sys.exit(main())
